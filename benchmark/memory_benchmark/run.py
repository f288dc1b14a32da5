#!/usr/bin/env python3
"""Operator memory benchmark replication (reference:
benchmark/memory_benchmark — RSS vs CR count vs pod count; conclusion there:
RSS correlates with POD count, not CR count).

Three experiments mirroring the reference:
  1. N head-only clusters        (CR-heavy, pod-light)
  2. 1 cluster scaled to N pods  (pod-heavy, CR-light)
  3. N/5 clusters x 5 pods       (mixed)
Prints RSS after each step; with the native C++ store the curve is flat.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def rss_mb():
    import psutil
    return psutil.Process().memory_info().rss / (1024 * 1024)


def measure(fn, *args):
    import gc
    gc.collect()
    before = rss_mb()
    out = fn(*args)
    gc.collect()
    return {"rss_before_mb": round(before, 1), "rss_after_mb": round(rss_mb(), 1),
            **out}


def experiment(total_pods, mode):
    from kuberay_amd.testing import ControlPlane, simple_raycluster
    cp = ControlPlane(kubelet_delay=0.0, record_events=False,
                      requeue_seconds=3600, poll_seconds=5.0)
    cp.start()
    try:
        if mode == "head-only":
            n_clusters, workers = total_pods, 0
        elif mode == "one-cluster":
            n_clusters, workers = 1, total_pods - 1
        else:
            n_clusters, workers = total_pods // 5, 4
        for i in range(n_clusters):
            cp.client.create(simple_raycluster(f"m-{i:04d}", workers=workers))
        deadline = time.monotonic() + 300
        while cp.server.count("Pod") < n_clusters * (workers + 1) and \
                time.monotonic() < deadline:
            time.sleep(0.05)
        time.sleep(1.0)
        backend_bytes = None
        backend = cp.server._backend
        if hasattr(backend, "total_bytes"):
            backend_bytes = backend.total_bytes()
        return {"mode": mode, "clusters": n_clusters,
                "pods": cp.server.count("Pod"),
                "store_bytes": backend_bytes}
    finally:
        cp.stop()


def main():
    results = []
    for mode in ("head-only", "one-cluster", "mixed"):
        for total in (50, 150):
            results.append(measure(experiment, total, mode))
            print(json.dumps(results[-1]))
    here = os.path.dirname(os.path.abspath(__file__))
    with open(os.path.join(here, "results.json"), "w") as f:
        json.dump(results, f, indent=2)


if __name__ == "__main__":
    main()
