"""Operator memory benchmark (reference analog:
benchmark/memory_benchmark/memory_benchmark.md — three experiments that
correlate operator-pod memory with the number of Ray pods, not CRs).

Methodology: one REAL standalone operator subprocess (memory backend, no
torch import) serving its kube-API facade; this script drives each
experiment over REST and samples the operator process RSS after each
addition converges. The reference paces experiments on GKE wall-clock
(20 s / 60 s per step); here each step is convergence-gated instead, so
a run finishes in minutes while exercising the identical object counts:

* exp1 — 150 head-only RayClusters (CR-heavy, pod-light)
* exp2 — 1 RayCluster scaled by +5 workers until 150 pods (pod-heavy)
* exp3 — 30 five-pod RayClusters (1 head + 4 workers)

Output: JSON lines per sample {experiment, n_crs, n_pods, rss_mb} to
results-memory.json, comparable to the reference's figure axes.
"""
import argparse
import json
import os
import subprocess
import sys
import time

import httpx
import psutil

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from kuberay_amd.kube.rest import RestClient  # noqa: E402
from kuberay_amd.models import RayCluster  # noqa: E402
from kuberay_amd.testing import simple_raycluster  # noqa: E402

PORT = 18443


def free_port() -> int:
    import socket
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        return sk.getsockname()[1]


def start_operator():
    env = dict(os.environ, PYTHONUNBUFFERED="1")
    proc = subprocess.Popen(
        [sys.executable, "-m", "kuberay_amd.operator",
         "--api-port", str(PORT), "--no-metrics"],
        env=env, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    for _ in range(100):
        try:
            httpx.get(f"http://127.0.0.1:{PORT}/api/v1/pods", timeout=1)
            return proc
        except httpx.HTTPError:
            time.sleep(0.2)
    proc.kill()
    raise RuntimeError("operator facade did not come up")


def rss_mb(proc) -> float:
    return psutil.Process(proc.pid).memory_info().rss / 1e6


def wait_ready(client, name, timeout=60):
    deadline = time.time() + timeout
    while time.time() < deadline:
        c = client.try_get(RayCluster, "default", name)
        if c is not None and c.status and c.status.state == "ready":
            return True
        time.sleep(0.1)
    return False


def n_pods(client) -> int:
    return len(client.raw_list("Pod", "default"))


def experiment1(client, proc, out, crs=150):
    """150 head-only clusters, sample every 10."""
    for i in range(crs):
        client.create(simple_raycluster(f"exp1-{i}", workers=0))
        if (i + 1) % 10 == 0:
            assert wait_ready(client, f"exp1-{i}")
            out.append({"experiment": 1, "n_crs": i + 1,
                        "n_pods": n_pods(client), "rss_mb": rss_mb(proc)})
    cleanup(client)


def experiment2(client, proc, out, target_pods=150):
    """One cluster, +5 workers per step until 150 pods."""
    client.create(simple_raycluster("exp2", workers=0))
    assert wait_ready(client, "exp2")
    replicas = 0
    while True:
        replicas += 5
        c = client.get(RayCluster, "default", "exp2")
        c.spec.worker_group_specs[0].replicas = replicas
        c.spec.worker_group_specs[0].max_replicas = max(replicas, 150)
        client.update(c)
        deadline = time.time() + 60
        while time.time() < deadline and n_pods(client) < 1 + replicas:
            time.sleep(0.1)
        pods = n_pods(client)
        out.append({"experiment": 2, "n_crs": 1, "n_pods": pods,
                    "rss_mb": rss_mb(proc)})
        if pods >= target_pods:
            break
    cleanup(client)


def experiment3(client, proc, out, crs=30):
    """30 five-pod clusters (1 head + 4 workers), sample every 5."""
    for i in range(crs):
        client.create(simple_raycluster(f"exp3-{i}", workers=4))
        if (i + 1) % 5 == 0:
            assert wait_ready(client, f"exp3-{i}")
            out.append({"experiment": 3, "n_crs": i + 1,
                        "n_pods": n_pods(client), "rss_mb": rss_mb(proc)})
    cleanup(client)


def cleanup(client):
    for c in client.list(RayCluster, "default"):
        client.delete(RayCluster, "default", c.metadata.name)
    deadline = time.time() + 60
    while time.time() < deadline and n_pods(client) > 0:
        time.sleep(0.2)


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default=os.path.join(
        os.path.dirname(__file__), "results-memory.json"))
    ap.add_argument("--port", type=int, default=0,
                    help="facade port (0 = pick a free one)")
    args = ap.parse_args()
    global PORT
    PORT = args.port or free_port()
    proc = start_operator()
    out = []
    try:
        client = RestClient(base_url=f"http://127.0.0.1:{PORT}")
        base = rss_mb(proc)
        out.append({"experiment": 0, "n_crs": 0, "n_pods": 0,
                    "rss_mb": base, "note": "idle baseline"})
        t0 = time.time()
        experiment1(client, proc, out)
        experiment2(client, proc, out)
        experiment3(client, proc, out)
        summary = {
            "baseline_rss_mb": round(base, 1),
            "exp1_peak_rss_mb": round(max(r["rss_mb"] for r in out
                                          if r["experiment"] == 1), 1),
            "exp2_peak_rss_mb": round(max(r["rss_mb"] for r in out
                                          if r["experiment"] == 2), 1),
            "exp3_peak_rss_mb": round(max(r["rss_mb"] for r in out
                                          if r["experiment"] == 3), 1),
            "wall_s": round(time.time() - t0, 1),
        }
        with open(args.out, "w") as f:
            json.dump({"summary": summary, "samples": out}, f, indent=1)
        print(json.dumps(summary))
    finally:
        proc.terminate()
        proc.wait(timeout=10)
    return 0


if __name__ == "__main__":
    sys.exit(main())
