#!/usr/bin/env python3
"""Sharded-operator topology bench (docs/roadmap.md item: sharded operators
as REAL separate processes against the kube-API facade).

Topology:
  parent process:  InMemoryApiServer + KubeApiFacade (HTTP) + SimKubelet
  N subprocesses:  `python -m kuberay_amd.operator --backend kubernetes
                    --kubeconfig <facade> --shards N --shard-index i`
                   each electing its own per-shard leader Lease and owning
                   the CRs whose crc32(ns/name) %% N == i.

Measures: clusters→Ready throughput end-to-end over HTTP, and verifies the
shard split by scraping each operator's /metrics for
kuberay_reconcile_total{controller="raycluster"}.

Usage: python benchmark/perf-tests/sharded.py --shards 2 --clusters 200
"""
from __future__ import annotations

import argparse
import json
import os
import re
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def write_kubeconfig(url: str) -> str:
    import yaml
    cfg = {
        "apiVersion": "v1", "kind": "Config", "current-context": "facade",
        "clusters": [{"name": "facade", "cluster": {"server": url}}],
        "users": [{"name": "facade", "user": {}}],
        "contexts": [{"name": "facade",
                      "context": {"cluster": "facade", "user": "facade"}}],
    }
    f = tempfile.NamedTemporaryFile("w", suffix=".kubeconfig", delete=False)
    yaml.safe_dump(cfg, f)
    f.close()
    return f.name


def scrape_reconciles(port: int) -> float:
    import httpx
    try:
        text = httpx.get(f"http://127.0.0.1:{port}/metrics", timeout=5).text
    except httpx.HTTPError:
        return 0.0
    total = 0.0
    for line in text.splitlines():
        m = re.match(r'kuberay_reconcile_total\{controller="raycluster",'
                     r'outcome="[^"]+"\} ([0-9.e+]+)', line)
        if m:
            total += float(m.group(1))
    return total


def main() -> int:
    parser = argparse.ArgumentParser()
    parser.add_argument("--shards", type=int, default=2)
    parser.add_argument("--clusters", type=int, default=200)
    parser.add_argument("--timeout", type=float, default=300.0)
    parser.add_argument("--churn-minutes", type=float, default=0.0,
                        help="after ready: sustained scale churn over REST")
    args = parser.parse_args()

    from kuberay_amd.kube.httpserver import KubeApiFacade
    from kuberay_amd.kube.kubelet import SimKubelet
    from kuberay_amd.kube.rest import RestClient
    from kuberay_amd.kube.store import InMemoryApiServer
    from kuberay_amd.testing import simple_raycluster

    store = InMemoryApiServer()
    facade = KubeApiFacade(store).start()
    kubelet = SimKubelet(store, startup_delay=0.0, executors=2)
    kubelet.start()
    kubeconfig = write_kubeconfig(facade.url)

    procs = []
    metric_ports = []
    repo = os.path.join(os.path.dirname(__file__), "..", "..")
    for i in range(args.shards):
        port = 18100 + i
        metric_ports.append(port)
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "kuberay_amd.operator",
             "--backend", "kubernetes", "--kubeconfig", kubeconfig,
             "--shards", str(args.shards), "--shard-index", str(i),
             "--metrics-addr", f"127.0.0.1:{port}"],
            cwd=repo, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL))

    client = RestClient(base_url=facade.url)
    result = {"suite": f"{args.shards}-shard-operator-topology",
              "shards": args.shards, "clusters": args.clusters}
    try:
        # wait for every shard's health endpoint (leader elected + started)
        import httpx
        deadline = time.monotonic() + 60
        for port in metric_ports:
            while time.monotonic() < deadline:
                try:
                    if httpx.get(f"http://127.0.0.1:{port}/healthz",
                                 timeout=2).status_code == 200:
                        break
                except httpx.HTTPError:
                    time.sleep(0.2)
            else:
                raise RuntimeError(f"shard on :{port} never became healthy")

        watcher = store.watch({"RayCluster"})
        t0 = time.perf_counter()
        names = set()
        for i in range(args.clusters):
            name = f"shardbench-{i:04d}"
            client.create(simple_raycluster(name, workers=3,
                                            gpus_per_worker=1))
            names.add(name)
        pending = set(names)
        stop_at = time.monotonic() + args.timeout
        while pending and time.monotonic() < stop_at:
            ev = watcher.next(timeout=0.25)
            if ev is None:
                continue
            _, obj = ev
            if obj["metadata"]["name"] in pending and \
                    obj.get("status", {}).get("state") == "ready":
                pending.discard(obj["metadata"]["name"])
        elapsed = time.perf_counter() - t0
        watcher.stop()
        result["all_ready"] = not pending
        result["ready_wait_s"] = round(elapsed, 2)
        result["clusters_per_s"] = round(args.clusters / elapsed, 1)
        per_shard = [scrape_reconciles(p) for p in metric_ports]
        result["reconciles_per_shard"] = per_shard
        # every shard must have done real work (the CR space is hash-split)
        result["all_shards_active"] = all(c > 0 for c in per_shard)

        if args.churn_minutes > 0:
            # sustained scale churn over REST: read-modify-write with
            # conflict retry against live sharded operators, then pin every
            # cluster to a known shape and require full convergence
            import random
            import threading

            from kuberay_amd.kube.store import NotFoundError
            from kuberay_amd.models import RayCluster
            ordered = sorted(names)
            stop = threading.Event()
            counts = {"ops": 0, "errors": 0}
            lock = threading.Lock()

            def churn(seed):
                rng = random.Random(seed)
                local = RestClient(base_url=facade.url)
                while not stop.is_set():
                    name = rng.choice(ordered)
                    try:
                        def mutate(rc):
                            rc.spec.worker_group_specs[0].replicas = \
                                rng.randrange(0, 5)
                        local.update_with_retry(RayCluster, "default",
                                                name, mutate, attempts=20)
                        with lock:
                            counts["ops"] += 1
                    except NotFoundError:
                        pass
                    except Exception:  # noqa: BLE001
                        with lock:
                            counts["errors"] += 1
                    time.sleep(0.01)

            churners = [threading.Thread(target=churn, args=(s,),
                                         daemon=True) for s in range(4)]
            for t in churners:
                t.start()
            time.sleep(args.churn_minutes * 60)
            stop.set()
            for t in churners:
                t.join(timeout=5)
            # pin and require convergence
            for name in ordered:
                def pin(rc):
                    rc.spec.worker_group_specs[0].replicas = 1
                client.update_with_retry(RayCluster, "default", name, pin,
                                         attempts=50)
            deadline = time.monotonic() + 120

            def all_converged():
                for name in ordered:
                    obj = store.try_get("RayCluster", "default", name)
                    st = (obj or {}).get("status", {})
                    if st.get("state") != "ready" or \
                            st.get("availableWorkerReplicas") != 1:
                        return False
                return True
            converged = False
            while time.monotonic() < deadline:
                if all_converged():
                    converged = True
                    break
                time.sleep(0.5)
            result["churn_minutes"] = args.churn_minutes
            result["churn_ops"] = counts["ops"]
            result["churn_client_errors"] = counts["errors"]
            result["churn_converged"] = converged
            result["reconciles_per_shard_after_churn"] = [
                scrape_reconciles(p) for p in metric_ports]
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        kubelet.stop()
        facade.stop()
        os.unlink(kubeconfig)

    print(json.dumps(result, indent=2))
    ok = result.get("all_ready") and result.get("all_shards_active")
    if args.churn_minutes > 0:
        ok = ok and result.get("churn_converged") and \
            result.get("churn_client_errors") == 0
    return 0 if ok else 1


if __name__ == "__main__":
    raise SystemExit(main())
