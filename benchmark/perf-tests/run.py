#!/usr/bin/env python3
"""Perf-test suite replication (reference: benchmark/perf-tests clusterloader2
configs at 100/1k/5k/10k scale).

Runs the RayCluster and RayJob lifecycle measurements at each scale tier
against the in-process control plane (the reference's numbers were measured
on GKE with real kubelets; ours use the simulated kubelet — the envtest
methodology — so compare operator-side steps, not pod cold-starts).

Usage: python benchmark/perf-tests/run.py [--scale 100|500|1000|5000]
Writes junit-style XML + JSON results next to this script.
"""
import argparse
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

SCALES = {
    "100": {"clusters": 100, "namespaces": 10},
    "500": {"clusters": 500, "namespaces": 10},
    "1000": {"clusters": 1000, "namespaces": 20},
    "5000": {"clusters": 5000, "namespaces": 50},
    "10000": {"clusters": 10000, "namespaces": 100},
}

REFERENCE = {  # junit totals from the reference repo (GKE, v1.1.1)
    "100": 135.2, "1000": 647.5, "5000": 2700.1, "10000": 2720.1,
}
REFERENCE_RAYJOB = {"100": 712.8, "1000": 997.2, "5000": 3059.0,
                    "10000": 3459.0}


def run_raycluster_tier(scale_cfg, workers=4):
    from kuberay_amd.testing import ControlPlane, simple_raycluster
    from kuberay_amd.models import RayCluster
    n = scale_cfg["clusters"]
    namespaces = [f"perf-{i}" for i in range(scale_cfg["namespaces"])]
    cp = ControlPlane(kubelet_delay=0.0, record_events=False,
                      requeue_seconds=3600, poll_seconds=5.0, workers=workers)
    cp.start()
    try:
        watcher = cp.server.watch({"RayCluster"})
        t0 = time.perf_counter()
        for i in range(n):
            cp.client.create(simple_raycluster(
                f"perf-{i:05d}", namespace=namespaces[i % len(namespaces)],
                workers=3))
        t_create = time.perf_counter() - t0
        pending = {f"perf-{i:05d}" for i in range(n)}
        while pending:
            ev = watcher.next(timeout=1.0)
            if ev is None:
                continue
            _, obj = ev
            if obj.get("status", {}).get("state") == "ready":
                pending.discard(obj["metadata"]["name"])
        t_ready = time.perf_counter() - t0
        watcher.stop()
        for i in range(n):
            cp.server.delete("RayCluster", namespaces[i % len(namespaces)],
                             f"perf-{i:05d}")
        while cp.server.count("RayCluster"):
            time.sleep(0.01)
        t_total = time.perf_counter() - t0
    finally:
        cp.stop()
    return {"create_s": round(t_create, 2), "ready_wait_s": round(t_ready, 2),
            "overall_s": round(t_total, 2)}


def run_rayjob_tier(scale_cfg, workers=4):
    """N RayJobs (ephemeral clusters, shutdown on finish) to Complete —
    the reference's N-rayjob clusterloader2 suite."""
    from kuberay_amd.models import RayJob
    from kuberay_amd.testing import ControlPlane, simple_raycluster
    n = scale_cfg["clusters"]
    cp = ControlPlane(kubelet_delay=0.0, record_events=False,
                      requeue_seconds=3600, poll_seconds=0.2, workers=workers,
                      job_runtime=0.01)
    cp.dashboard.job_polls_to_succeeded = 1
    cp.start()
    try:
        watcher = cp.server.watch({"RayJob"})
        t0 = time.perf_counter()
        for i in range(n):
            cp.client.create(RayJob.from_dict({
                "apiVersion": "ray.io/v1", "kind": "RayJob",
                "metadata": {"name": f"perfjob-{i:05d}", "namespace": "default"},
                "spec": {"entrypoint": "python train.py",
                         "shutdownAfterJobFinishes": True,
                         "rayClusterSpec": simple_raycluster(
                             "x", workers=1).spec.to_dict()}}))
        t_create = time.perf_counter() - t0
        pending = {f"perfjob-{i:05d}" for i in range(n)}
        while pending:
            ev = watcher.next(timeout=1.0)
            if ev is None:
                continue
            _, obj = ev
            if obj.get("status", {}).get("jobDeploymentStatus") == "Complete":
                pending.discard(obj["metadata"]["name"])
        t_complete = time.perf_counter() - t0
        watcher.stop()
    finally:
        cp.stop()
    return {"create_s": round(t_create, 2),
            "jobs_complete_s": round(t_complete, 2),
            "overall_s": round(t_complete, 2)}


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--scale", default="100", choices=sorted(SCALES))
    parser.add_argument("--suite", default="raycluster",
                        choices=["raycluster", "rayjob"])
    args = parser.parse_args()
    if args.suite == "rayjob":
        result = run_rayjob_tier(SCALES[args.scale])
        ref = REFERENCE_RAYJOB.get(args.scale)
    else:
        result = run_raycluster_tier(SCALES[args.scale])
        ref = REFERENCE.get(args.scale)
    out = {
        "suite": f"{args.scale}-{args.suite}",
        "kubelet": "simulated",
        "note": ("rayjob suite uses synthetic instant applications: this "
                 "measures operator lifecycle overhead; the reference's "
                 "numbers include real MNIST workload runtime"
                 if args.suite == "rayjob" else
                 "measures operator-side steps; reference numbers include "
                 "real GKE pod cold-starts"),
        **result,
        "reference_overall_s_gke": ref,
        "speedup_vs_reference": round(ref / result["overall_s"], 1) if ref else None,
    }
    print(json.dumps(out, indent=2))
    here = os.path.dirname(os.path.abspath(__file__))
    with open(os.path.join(here, f"results-{args.scale}-{args.suite}.json"), "w") as f:
        json.dump(out, f, indent=2)
    # junit for parity with the reference artifact format
    wait_name = ("Wait for RayClusters ready" if args.suite == "raycluster"
                 else "Wait for RayJobs complete")
    wait_time = result.get("ready_wait_s", result.get("jobs_complete_s"))
    xml = (f'<testsuite name="{args.scale}-{args.suite}" tests="3">'
           f'<testcase name="overall" time="{result["overall_s"]}"/>'
           f'<testcase name="create" time="{result["create_s"]}"/>'
           f'<testcase name="{wait_name}" time="{wait_time}"/>'
           f'</testsuite>')
    with open(os.path.join(here, f"junit-{args.scale}-{args.suite}.xml"), "w") as f:
        f.write(xml)


if __name__ == "__main__":
    main()
