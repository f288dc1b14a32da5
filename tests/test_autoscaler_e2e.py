"""Autoscaler e2e through the full control plane (reference analog:
test/e2eautoscaler — drive scale-up/down and assert DesiredWorkerReplicas).

The rocm-smi telemetry is faked (synthetic actor load); everything else —
CR patch → watch → reconcile → pod create/delete → status — is the real
path, including the no-random-delete guarantee (victims must be named).
"""
import time

import pytest

from kuberay_amd.gpu.autoscaler import (
    AMD_AUTOSCALER_ANNOTATION,
    AutoscalerPolicy,
    MI355XAutoscaler,
)
from kuberay_amd.models import RayCluster
from kuberay_amd.testing import ControlPlane, simple_raycluster
from kuberay_amd.utils import constants as C


@pytest.fixture()
def stack():
    cp = ControlPlane(kubelet_delay=0.01, poll_seconds=0.05)
    cp.start()
    telemetry = {"avg_utilization_pct": 0.0, "max_utilization_pct": 0.0,
                 "avg_vram_used_fraction": 0.0, "max_vram_used_fraction": 0.0,
                 "gpu_count": 8}
    autoscaler = MI355XAutoscaler(
        cp.client, telemetry=lambda: dict(telemetry),
        policy=AutoscalerPolicy(up_stable_s=0.05, idle_timeout_s=0.1,
                                cooldown_s=0.05))
    yield cp, autoscaler, telemetry
    cp.stop()


def cluster_of(cp):
    return cp.client.get(RayCluster, "default", "demo")


class TestAutoscalerEndToEnd:
    def test_scale_up_under_synthetic_load(self, stack):
        cp, autoscaler, telemetry = stack
        cluster = simple_raycluster("demo", workers=0, gpus_per_worker=1,
                                    enableInTreeAutoscaling=True)
        cluster.spec.worker_group_specs[0].max_replicas = 8
        cluster.metadata.annotations = {AMD_AUTOSCALER_ANNOTATION: "true"}
        cp.client.create(cluster)
        assert cp.wait_for(
            lambda: cluster_of(cp).status.desired_worker_replicas == 0)

        # synthetic actor load saturates the GPUs; tick until the operator
        # has materialized >= 2 workers (each up-step needs a fresh
        # stability window, like the real 5s loop)
        telemetry["avg_utilization_pct"] = 95.0
        telemetry["max_vram_used_fraction"] = 0.9
        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            autoscaler.step()
            if cluster_of(cp).status.available_worker_replicas >= 2:
                break
            time.sleep(0.08)
        assert cluster_of(cp).status.available_worker_replicas >= 2
        assert cluster_of(cp).status.desired_worker_replicas >= 2

    def test_scale_down_when_idle_names_victims(self, stack):
        cp, autoscaler, telemetry = stack
        cluster = simple_raycluster("demo", workers=3, gpus_per_worker=1,
                                    enableInTreeAutoscaling=True)
        cluster.metadata.annotations = {AMD_AUTOSCALER_ANNOTATION: "true"}
        cp.client.create(cluster)
        assert cp.wait_for(
            lambda: cluster_of(cp).status.available_worker_replicas == 3,
            timeout=15)

        telemetry["avg_utilization_pct"] = 1.0
        telemetry["max_vram_used_fraction"] = 0.01
        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            autoscaler.step()
            if cluster_of(cp).status.available_worker_replicas <= 2:
                break
            time.sleep(0.06)
        rc = cluster_of(cp)
        assert rc.status.available_worker_replicas <= 2
        # the operator must not have random-deleted: decisions named victims
        # (WorkersToDelete was used and honored)
        assert rc.spec.worker_group_specs[0].replicas <= 2

    def test_in_tree_sidecar_contract_still_respected(self, stack):
        """A (simulated) in-pod Ray autoscaler patching Replicas +
        WorkersToDelete composes with the rocm-smi loop (§3.4 contract)."""
        cp, autoscaler, telemetry = stack
        cluster = simple_raycluster("demo", workers=2, gpus_per_worker=1,
                                    enableInTreeAutoscaling=True)
        cp.client.create(cluster)
        assert cp.wait_for(
            lambda: cluster_of(cp).status.available_worker_replicas == 2,
            timeout=15)
        victim = next(
            v.name for v in cp.client.list_pod_views(
                "default", {C.RAY_CLUSTER_LABEL_KEY: "demo",
                            C.RAY_NODE_TYPE_LABEL_KEY: "worker"}))
        # what Ray's sidecar does: PATCH replicas + workersToDelete
        cp.client.patch(RayCluster, "default", "demo", {"spec": {
            "workerGroupSpecs": [
                {**cluster_of(cp).spec.worker_group_specs[0].to_dict(),
                 "replicas": 1,
                 "scaleStrategy": {"workersToDelete": [victim]}}]}})
        assert cp.wait_for(
            lambda: cluster_of(cp).status.available_worker_replicas == 1,
            timeout=15)
        names = [v.name for v in cp.client.list_pod_views(
            "default", {C.RAY_CLUSTER_LABEL_KEY: "demo",
                        C.RAY_NODE_TYPE_LABEL_KEY: "worker"})]
        assert victim not in names


class TestZeroToEight:
    def test_full_0_to_8_scale_up_then_drain(self, stack):
        """BASELINE config #5 shape: 0 -> 8 MI355X workers under sustained
        synthetic actor load, then a full drain back to min when idle."""
        cp, autoscaler, telemetry = stack
        cluster = simple_raycluster("demo", workers=0, gpus_per_worker=1,
                                    enableInTreeAutoscaling=True)
        cluster.spec.worker_group_specs[0].min_replicas = 0
        cluster.spec.worker_group_specs[0].max_replicas = 8
        cluster.metadata.annotations = {AMD_AUTOSCALER_ANNOTATION: "true"}
        cp.client.create(cluster)
        assert cp.wait_for(lambda: cluster_of(cp).status.state == "ready")

        telemetry["avg_utilization_pct"] = 95.0
        telemetry["max_vram_used_fraction"] = 0.9
        deadline = time.monotonic() + 40
        while time.monotonic() < deadline:
            autoscaler.step()
            if cluster_of(cp).status.available_worker_replicas == 8:
                break
            time.sleep(0.08)
        assert cluster_of(cp).status.available_worker_replicas == 8
        # never exceeds maxReplicas
        autoscaler.step()
        assert cluster_of(cp).status.desired_worker_replicas == 8

        # load drops: drain back toward min with NAMED victims each step
        telemetry["avg_utilization_pct"] = 2.0
        telemetry["max_vram_used_fraction"] = 0.02
        deadline = time.monotonic() + 40
        while time.monotonic() < deadline:
            autoscaler.step()
            if cluster_of(cp).status.available_worker_replicas == 0:
                break
            time.sleep(0.12)
        assert cluster_of(cp).status.available_worker_replicas == 0
        assert cluster_of(cp).status.desired_worker_replicas == 0


class TestPolicyMechanics:
    """Pure policy-level tests with a fake clock and telemetry: the
    stability-window / cooldown / clamp rules that keep rocm-smi-driven
    scaling from flapping."""

    def _mk(self, util=90.0, hbm=0.2):
        from kuberay_amd.gpu.autoscaler import (AMD_AUTOSCALER_ANNOTATION,
                                                AutoscalerPolicy,
                                                MI355XAutoscaler)
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.testing import simple_raycluster
        client = InMemoryClient()
        c = simple_raycluster("as", workers=2, gpus_per_worker=1)
        c.metadata.annotations = {AMD_AUTOSCALER_ANNOTATION: "true"}
        c.spec.worker_group_specs[0].min_replicas = 1
        c.spec.worker_group_specs[0].max_replicas = 3
        client.create(c)
        self.now = 1000.0
        self.summary = {"avg_utilization_pct": util,
                        "max_vram_used_fraction": hbm}
        policy = AutoscalerPolicy(up_stable_s=10, idle_timeout_s=60,
                                  cooldown_s=30)
        a = MI355XAutoscaler(client, telemetry=lambda: dict(self.summary),
                             policy=policy, clock=lambda: self.now)
        return a, client

    def _replicas(self, client):
        from kuberay_amd.models import RayCluster
        c = client.get(RayCluster, "default", "as")
        return c.spec.worker_group_specs[0].replicas

    def test_stability_window_gates_scale_up(self):
        a, client = self._mk(util=95)
        assert a.step() == []            # first sighting starts the window
        self.now += 5
        assert a.step() == []            # still inside up_stable_s
        self.now += 6
        assert any("scale-up" in d for d in a.step())
        assert self._replicas(client) == 3

    def test_cooldown_limits_action_rate(self):
        a, client = self._mk(util=95)
        a.step(); self.now += 11; a.step()
        assert self._replicas(client) == 3
        # high again immediately: cooldown (30 s) must hold even after
        # another full stability window
        self.now += 12
        assert a.step() == []
        assert self._replicas(client) == 3

    def test_max_replicas_clamps(self):
        a, client = self._mk(util=95)
        for _ in range(6):
            a.step(); self.now += 50
        assert self._replicas(client) == 3  # maxReplicas

    def test_min_replicas_clamps_scale_down(self):
        a, client = self._mk(util=1, hbm=0.05)
        for _ in range(8):
            a.step(); self.now += 100
        assert self._replicas(client) == 1  # minReplicas floor

    def test_flapping_utilization_never_acts(self):
        a, client = self._mk()
        for i in range(20):
            self.summary["avg_utilization_pct"] = 95 if i % 2 == 0 else 40
            assert a.step() == []
            self.now += 6   # window resets before up_stable_s elapses
        assert self._replicas(client) == 2

    def test_suspended_cluster_skipped(self):
        from kuberay_amd.models import RayCluster
        a, client = self._mk(util=95)
        def suspend(c):
            c.spec.suspend = True
        client.update_with_retry(RayCluster, "default", "as", suspend)
        for _ in range(4):
            assert a.step() == []
            self.now += 50
        assert self._replicas(client) == 2
