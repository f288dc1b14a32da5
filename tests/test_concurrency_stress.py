"""Concurrency stress (the reference's `go test -race` analog): many client
threads mutate clusters while all controllers run; the system must converge
with zero reconcile errors and consistent end state."""
import random
import threading
import time

import pytest

from kuberay_amd.models import RayCluster
from kuberay_amd.testing import ControlPlane, simple_raycluster


@pytest.mark.timeout(120)
class TestConcurrencyStress:
    def test_parallel_churn_converges_clean(self):
        cp = ControlPlane(kubelet_delay=0.0, poll_seconds=0.05,
                          record_events=False)
        cp.start()
        errors = []
        N_CLUSTERS = 12
        N_THREADS = 6
        OPS_PER_THREAD = 40
        try:
            for i in range(N_CLUSTERS):
                cp.client.create(simple_raycluster(f"st-{i}", workers=1))

            rng_global = random.Random(1234)
            seeds = [rng_global.randrange(1 << 30) for _ in range(N_THREADS)]

            def churn(seed):
                rng = random.Random(seed)
                for _ in range(OPS_PER_THREAD):
                    name = f"st-{rng.randrange(N_CLUSTERS)}"
                    try:
                        op = rng.random()
                        rc = cp.client.try_get(RayCluster, "default", name)
                        if rc is None:
                            continue
                        if op < 0.5:
                            rc.spec.worker_group_specs[0].replicas = \
                                rng.randrange(0, 4)
                            cp.client.update(rc)
                        elif op < 0.7:
                            rc.spec.suspend = not rc.spec.suspend
                            cp.client.update(rc)
                        else:
                            cp.client.patch(
                                RayCluster, "default", name,
                                {"metadata": {"annotations": {
                                    "stress": str(rng.random())}}})
                    except Exception as e:  # conflicts are expected; real
                        from kuberay_amd.kube.store import ApiError
                        if not isinstance(e, ApiError):
                            errors.append(e)
                    time.sleep(rng.random() * 0.01)

            threads = [threading.Thread(target=churn, args=(s,)) for s in seeds]
            for t in threads:
                t.start()
            for t in threads:
                t.join(timeout=60)

            assert not errors, errors[:3]

            # un-suspend everything and pin final replica counts
            for i in range(N_CLUSTERS):
                while True:
                    rc = cp.client.get(RayCluster, "default", f"st-{i}")
                    rc.spec.suspend = False
                    rc.spec.worker_group_specs[0].replicas = 1
                    try:
                        cp.client.update(rc)
                        break
                    except Exception:
                        time.sleep(0.01)

            def all_converged():
                for i in range(N_CLUSTERS):
                    rc = cp.client.get(RayCluster, "default", f"st-{i}")
                    if rc.status.state != "ready":
                        return False
                    if rc.status.available_worker_replicas != 1:
                        return False
                return True
            assert cp.wait_for(all_converged, timeout=60), [
                (cp.client.get(RayCluster, "default", f"st-{i}").status.state,
                 cp.client.get(RayCluster, "default",
                               f"st-{i}").status.available_worker_replicas)
                for i in range(N_CLUSTERS)]

            # no controller saw an unhandled exception
            for c in cp.manager.controllers:
                assert c.error_count == 0, c.name
            # exactly one head + one worker per cluster — no pod leaks
            for i in range(N_CLUSTERS):
                views = cp.client.list_pod_views(
                    "default", {"ray.io/cluster": f"st-{i}"})
                live = [v for v in views if not v.deletion_timestamp]
                assert len(live) == 2, (f"st-{i}", [v.name for v in live])
        finally:
            cp.stop()
