#!/usr/bin/env python3
"""RayService zero-downtime upgrade storm (BASELINE config #4 at scale).

Creates N RayServices, waits for Running, then flips every service's
worker image at once — forcing N simultaneous zero-downtime upgrades
(pending cluster per service -> serve config -> promote -> old cluster
GC). Asserts: every service ends Running on a NEW cluster, and no service
ever reported itself unready during its upgrade window (the zero-downtime
invariant, sampled continuously).

Usage: python benchmark/perf-tests/rayservice_upgrade.py --services 50
"""
import argparse
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

SERVE_CONFIG = "applications:\n- name: app1\n  import_path: m.g\n"


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--services", type=int, default=50)
    parser.add_argument("--timeout", type=float, default=300.0)
    parser.add_argument("--incremental", action="store_true",
                        help="NewClusterWithIncrementalUpgrade via Gateway "
                             "API weighted migration (25%% steps)")
    args = parser.parse_args()

    from kuberay_amd.models import RayService
    from kuberay_amd.testing import ControlPlane, simple_raycluster

    if args.incremental:
        import kuberay_amd.features as features
        features.set_gate("RayServiceIncrementalUpgrade", True)

    cp = ControlPlane(kubelet_delay=0.0, record_events=False,
                      poll_seconds=0.2, workers=4)
    cp.rayservice_reconciler.cluster_deletion_delay_s = 1.0
    cp.start()
    t0 = time.monotonic()
    try:
        names = [f"upsvc-{i:03d}" for i in range(args.services)]
        # incremental upgrades require the autoscaler + intervalSeconds > 0
        # (ValidateClusterUpgradeOptions, validation.go:718-750)
        upgrade_spec = ({"upgradeStrategy": {
            "type": "NewClusterWithIncrementalUpgrade",
            "clusterUpgradeOptions": {"gatewayClassName": "istio",
                                      "stepSizePercent": 25,
                                      "intervalSeconds": 1,
                                      "maxSurgePercent": 100}}}
            if args.incremental else {})
        for name in names:
            cluster_spec = simple_raycluster(
                "x", workers=1, gpus_per_worker=1).spec.to_dict()
            if args.incremental:
                cluster_spec["enableInTreeAutoscaling"] = True
            cp.client.create(RayService.from_dict({
                "apiVersion": "ray.io/v1", "kind": "RayService",
                "metadata": {"name": name, "namespace": "default"},
                "spec": {"serveConfigV2": SERVE_CONFIG,
                         "rayClusterConfig": cluster_spec,
                         **upgrade_spec}}))

        def state_of(name):
            svc = cp.client.try_get(RayService, "default", name)
            if svc is None:
                return None, None
            return (svc.status.service_status,
                    svc.status.active_service_status.ray_cluster_name)

        def all_running():
            return all(state_of(n)[0] == "Running" and state_of(n)[1]
                       for n in names)

        deadline = time.monotonic() + args.timeout
        while not all_running() and time.monotonic() < deadline:
            time.sleep(0.3)
        assert all_running(), "initial provisioning did not converge"
        provision_s = time.monotonic() - t0
        first_clusters = {n: state_of(n)[1] for n in names}

        # continuous zero-downtime sampler: no service may leave Running
        violations = []
        stop = threading.Event()

        def sampler():
            while not stop.is_set():
                for n in names:
                    st, _ = state_of(n)
                    if st not in ("Running",):
                        violations.append((n, st, time.monotonic()))
                time.sleep(0.2)

        sampler_thread = threading.Thread(target=sampler, daemon=True)
        sampler_thread.start()

        # the storm: flip every worker image simultaneously
        t1 = time.monotonic()
        for n in names:
            while True:
                svc = cp.client.get(RayService, "default", n)
                svc.spec.ray_cluster_spec.worker_group_specs[0].template \
                    .spec.containers[0].image = "rocm/ray:2.47.0-upgraded"
                try:
                    cp.client.update(svc)
                    break
                except Exception:
                    time.sleep(0.02)

        def all_promoted():
            return all(state_of(n)[0] == "Running"
                       and state_of(n)[1] not in (None, first_clusters[n])
                       for n in names)

        deadline = time.monotonic() + args.timeout
        while not all_promoted() and time.monotonic() < deadline:
            time.sleep(0.3)
        promoted = all_promoted()
        upgrade_s = time.monotonic() - t1
        stop.set()
        sampler_thread.join(timeout=5)

        errors = sum(c.error_count for c in cp.manager.controllers)
        result = {
            "suite": f"{args.services}-rayservice-upgrade-storm"
                     + ("-incremental" if args.incremental else ""),
            "provision_s": round(provision_s, 2),
            "upgrade_storm_s": round(upgrade_s, 2),
            "all_promoted": promoted,
            "zero_downtime_violations": len(violations),
            "reconcile_errors": errors,
        }
        print(json.dumps(result, indent=2))
        return 0 if promoted and not violations and not errors else 1
    finally:
        cp.stop()


if __name__ == "__main__":
    sys.exit(main())
