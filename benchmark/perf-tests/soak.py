#!/usr/bin/env python3
"""500-RayCluster sustained soak (BASELINE config #5).

Holds 500 RayClusters live while continuously churning for --minutes:
random scale-ups/downs, suspend/resume cycles, pod kills (fault injection),
and rocm-smi-driven autoscaler ticks with synthetic load. At the end, pins
every cluster to a known shape and asserts full convergence, zero
unhandled reconcile errors, and stable RSS.

Usage: python benchmark/perf-tests/soak.py --minutes 5 --clusters 500
"""
import argparse
import json
import os
import random
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def rss_mb():
    import psutil
    return psutil.Process().memory_info().rss / (1024 * 1024)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--minutes", type=float, default=5.0)
    parser.add_argument("--clusters", type=int, default=500)
    parser.add_argument("--churn-threads", type=int, default=4)
    parser.add_argument("--seed", type=int, default=42)
    args = parser.parse_args()

    from kuberay_amd.gpu.autoscaler import (
        AMD_AUTOSCALER_ANNOTATION,
        AutoscalerPolicy,
        MI355XAutoscaler,
    )
    from kuberay_amd.models import RayCluster
    from kuberay_amd.testing import ControlPlane, simple_raycluster

    cp = ControlPlane(kubelet_delay=0.0, record_events=False,
                      requeue_seconds=300, poll_seconds=1.0, workers=4)
    cp.start()
    t_start = time.monotonic()
    stop_at = t_start + args.minutes * 60

    # synthetic actor load oscillates so the autoscaler exercises both ways
    def telemetry():
        phase = (time.monotonic() - t_start) % 120
        high = phase < 60
        return {"gpu_count": 8,
                "avg_utilization_pct": 90.0 if high else 5.0,
                "max_utilization_pct": 95.0 if high else 8.0,
                "avg_vram_used_fraction": 0.8 if high else 0.05,
                "max_vram_used_fraction": 0.9 if high else 0.08}

    autoscaler = MI355XAutoscaler(
        cp.client, telemetry=telemetry,
        policy=AutoscalerPolicy(up_stable_s=5, idle_timeout_s=20, cooldown_s=10))

    stats = {"ops": 0, "client_errors": 0}
    try:
        print(f"creating {args.clusters} clusters...", flush=True)
        for i in range(args.clusters):
            rc = simple_raycluster(f"soak-{i:04d}", workers=1, gpus_per_worker=1)
            if i % 10 == 0:
                rc.metadata.annotations = {AMD_AUTOSCALER_ANNOTATION: "true"}
            cp.client.create(rc)

        def all_ready(expect_workers=None):
            for i in range(args.clusters):
                rc = cp.client.try_get(RayCluster, "default", f"soak-{i:04d}")
                if rc is None or rc.status.state != "ready":
                    return False
                if expect_workers is not None and \
                        rc.status.available_worker_replicas != expect_workers:
                    return False
            return True

        deadline = time.monotonic() + 300
        while not all_ready() and time.monotonic() < deadline:
            time.sleep(0.5)
        assert all_ready(), "initial provisioning did not converge"
        rss_baseline = rss_mb()
        print(f"all {args.clusters} ready in "
              f"{time.monotonic() - t_start:.1f}s; RSS {rss_baseline:.0f}MB",
              flush=True)

        stop = threading.Event()

        def churn(seed):
            rng = random.Random(seed)
            from kuberay_amd.kube.store import ApiError
            while not stop.is_set():
                i = rng.randrange(args.clusters)
                name = f"soak-{i:04d}"
                try:
                    roll = rng.random()
                    if roll < 0.35:  # scale
                        rc = cp.client.try_get(RayCluster, "default", name)
                        if rc:
                            rc.spec.worker_group_specs[0].replicas = \
                                rng.randrange(0, 4)
                            cp.client.update(rc)
                    elif roll < 0.5:  # suspend/resume cycle
                        rc = cp.client.try_get(RayCluster, "default", name)
                        if rc:
                            rc.spec.suspend = not rc.spec.suspend
                            cp.client.update(rc)
                    elif roll < 0.8:  # fault injection: kill a random pod
                        views = cp.client.list_pod_views(
                            "default", {"ray.io/cluster": name})
                        live = [v for v in views if not v.deletion_timestamp]
                        if live:
                            victim = rng.choice(live)
                            cp.server.patch_merge(
                                "Pod", "default", victim.name,
                                {"status": {"phase": "Failed"}},
                                subresource="status")
                    # else: just read (list pressure)
                    else:
                        cp.client.list_pod_views("default",
                                                 {"ray.io/cluster": name})
                    stats["ops"] += 1
                except ApiError:
                    pass
                except Exception:
                    stats["client_errors"] += 1
                time.sleep(rng.random() * 0.05)

        threads = [threading.Thread(target=churn, args=(args.seed + t,),
                                    daemon=True)
                   for t in range(args.churn_threads)]
        for t in threads:
            t.start()

        ticks = 0
        while time.monotonic() < stop_at:
            autoscaler.step()
            ticks += 1
            time.sleep(2.0)
        stop.set()
        for t in threads:
            t.join(timeout=10)
        print(f"churn done: {stats['ops']} ops, {ticks} autoscaler ticks",
              flush=True)

        # pin the final shape and require convergence
        for i in range(args.clusters):
            while True:
                rc = cp.client.get(RayCluster, "default", f"soak-{i:04d}")
                rc.spec.suspend = False
                rc.spec.worker_group_specs[0].replicas = 1
                rc.spec.worker_group_specs[0].scale_strategy.workers_to_delete = None
                try:
                    cp.client.update(rc)
                    break
                except Exception:
                    time.sleep(0.01)
        deadline = time.monotonic() + 300
        while not all_ready(expect_workers=1) and time.monotonic() < deadline:
            time.sleep(0.5)
        converged = all_ready(expect_workers=1)
        errors = sum(c.error_count for c in cp.manager.controllers)
        reconciles = sum(c.reconcile_count for c in cp.manager.controllers)
        rss_end = rss_mb()
        store_bytes = getattr(cp.server._backend, "total_bytes", lambda: -1)()
        result = {
            "suite": "500-raycluster-soak",
            "minutes": args.minutes,
            "clusters": args.clusters,
            "churn_ops": stats["ops"],
            "client_errors": stats["client_errors"],
            "reconciles": reconciles,
            "reconcile_errors": errors,
            "converged": converged,
            "rss_mb_baseline": round(rss_baseline, 1),
            "rss_mb_end": round(rss_end, 1),
            "rss_growth_mb": round(rss_end - rss_baseline, 1),
            "store_bytes_mb": round(store_bytes / 1e6, 2)
                              if store_bytes >= 0 else None,
        }
        print(json.dumps(result, indent=2))
        here = os.path.dirname(os.path.abspath(__file__))
        with open(os.path.join(here, "soak-results.json"), "w") as f:
            json.dump(result, f, indent=2)
        if not converged or errors:
            return 1
        return 0
    finally:
        cp.stop()


if __name__ == "__main__":
    sys.exit(main())
