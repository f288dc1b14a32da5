"""Incremental upgrade tests (reference analog: test/e2eincrementalupgrade +
rayservice_controller.go Gateway/HTTPRoute/TargetCapacity logic)."""
import time

import pytest

import kuberay_amd.features as features
from kuberay_amd.models import RayCluster, RayService
from kuberay_amd.testing import ControlPlane, simple_raycluster

SERVE_CONFIG = "applications:\n- name: app1\n  import_path: m.g\n"


def make_service(name="svc1", interval=1, step=50):
    # incremental upgrade requires the autoscaler (validation.go:719-721)
    cluster_spec = simple_raycluster("x", workers=1).spec.to_dict()
    cluster_spec["enableInTreeAutoscaling"] = True
    return RayService.from_dict({
        "apiVersion": "ray.io/v1", "kind": "RayService",
        "metadata": {"name": name, "namespace": "default"},
        "spec": {
            "serveConfigV2": SERVE_CONFIG,
            "rayClusterConfig": cluster_spec,
            "upgradeStrategy": {
                "type": "NewClusterWithIncrementalUpgrade",
                "clusterUpgradeOptions": {
                    "gatewayClassName": "istio",
                    "stepSizePercent": step,
                    "intervalSeconds": interval,
                    "maxSurgePercent": 100,
                }},
        },
    })


@pytest.fixture()
def cp():
    features.set_gate("RayServiceIncrementalUpgrade", True)
    plane = ControlPlane(kubelet_delay=0.01, poll_seconds=0.05)
    plane.rayservice_reconciler.cluster_deletion_delay_s = 0.2
    plane.start()
    yield plane
    plane.stop()
    features.reset()


def svc_of(cp, name="svc1"):
    return cp.client.get(RayService, "default", name)


def wait_ready(cp, name="svc1", timeout=25):
    return cp.wait_for(lambda: svc_of(cp, name).condition_true("Ready"), timeout)


def set_image_with_retry(cp, image, name="svc1"):
    def mutate(svc):
        svc.spec.ray_cluster_spec.worker_group_specs[0].template.spec \
            .containers[0].image = image
    cp.client.update_with_retry(RayService, "default", name, mutate)


def trigger_upgrade(cp):
    set_image_with_retry(cp, "rayproject/ray:2.47.0-rocm")


class TestIncrementalUpgrade:
    def test_gateway_and_route_created_with_weighted_migration(self, cp):
        cp.client.create(make_service())
        assert wait_ready(cp)
        old_active = svc_of(cp).status.active_service_status.ray_cluster_name
        trigger_upgrade(cp)

        # migration completes (2 steps of 50%) and promotes
        def promoted():
            s = svc_of(cp)
            return (s.status.active_service_status.ray_cluster_name
                    not in (None, old_active) and s.condition_true("Ready"))
        assert cp.wait_for(promoted, timeout=30)

        # gateway infra existed during migration and is cleaned after
        assert cp.wait_for(lambda: cp.server.count("HTTPRoute") == 0, timeout=10)
        assert cp.server.count("Gateway") == 0

    def test_traffic_percent_tracked_in_status(self, cp):
        cp.client.create(make_service(interval=3600, step=40))
        assert wait_ready(cp)
        trigger_upgrade(cp)

        def first_step():
            s = svc_of(cp)
            return (s.status.pending_service_status.traffic_routed_percent or 0) >= 40
        assert cp.wait_for(first_step, timeout=25)
        s = svc_of(cp)
        # interval=1h: held at the first step, not promoted
        assert s.status.pending_service_status.traffic_routed_percent == 40
        assert s.status.pending_service_status.last_traffic_migrated_time
        assert s.status.active_service_status.ray_cluster_name  # still active
        route = cp.server.list("HTTPRoute")[0]
        weights = {b["name"]: b["weight"]
                   for b in route["spec"]["rules"][0]["backendRefs"]}
        assert sorted(weights.values()) == [40, 60]

    def test_rollback_on_spec_revert(self, cp):
        cp.client.create(make_service(interval=3600, step=10))
        assert wait_ready(cp)
        original_image = svc_of(cp).spec.ray_cluster_spec \
            .worker_group_specs[0].template.spec.containers[0].image
        active = svc_of(cp).status.active_service_status.ray_cluster_name
        trigger_upgrade(cp)
        assert cp.wait_for(
            lambda: svc_of(cp).status.pending_service_status.ray_cluster_name,
            timeout=20)
        # revert the spec to the running cluster's shape mid-upgrade
        set_image_with_retry(cp, original_image)

        def rolled_back():
            s = svc_of(cp)
            return (not s.status.pending_service_status.ray_cluster_name
                    and s.status.active_service_status.ray_cluster_name == active)
        assert cp.wait_for(rolled_back, timeout=25)

    def test_gate_off_promotes_instantly(self):
        features.set_gate("RayServiceIncrementalUpgrade", False)
        plane = ControlPlane(kubelet_delay=0.01, poll_seconds=0.05)
        plane.rayservice_reconciler.cluster_deletion_delay_s = 0.2
        plane.start()
        try:
            plane.client.create(make_service())
            assert plane.wait_for(
                lambda: plane.client.get(RayService, "default", "svc1")
                .condition_true("Ready"), timeout=25)
            old = plane.client.get(RayService, "default", "svc1") \
                .status.active_service_status.ray_cluster_name

            def set_image(svc):
                svc.spec.ray_cluster_spec.worker_group_specs[0].template.spec \
                    .containers[0].image = "rayproject/ray:2.47.0-rocm"
            plane.client.update_with_retry(RayService, "default", "svc1",
                                           set_image)
            assert plane.wait_for(
                lambda: plane.client.get(RayService, "default", "svc1")
                .status.active_service_status.ray_cluster_name not in (None, old),
                timeout=25)
            assert plane.server.count("Gateway") == 0  # no gateway infra
        finally:
            plane.stop()
            features.reset()


class TestGatewayReadinessGate:
    """Parity: util.go:874-915 + rayservice_controller.go:1657-1667 —
    traffic weights only advance once the Gateway is Programmed and the
    HTTPRoute's parent status is Accepted + ResolvedRefs."""

    def test_readiness_helpers(self):
        from kuberay_amd.ops.incremental import (is_gateway_ready,
                                                 is_http_route_ready)
        gw = {"metadata": {"name": "g", "namespace": "default"},
              "status": {"conditions": [
                  {"type": "Accepted", "status": "True"},
                  {"type": "Programmed", "status": "True"}]}}
        assert is_gateway_ready(gw)
        assert not is_gateway_ready({"metadata": {"name": "g"},
                                     "status": {"conditions": [
                                         {"type": "Accepted",
                                          "status": "True"}]}})
        assert not is_gateway_ready(None)
        route = {"status": {"parents": [{
            "parentRef": {"name": "g"},
            "conditions": [{"type": "Accepted", "status": "True"},
                           {"type": "ResolvedRefs", "status": "True"}]}]}}
        assert is_http_route_ready(gw, route)
        # parent entry for a DIFFERENT gateway doesn't count
        other = {"status": {"parents": [{
            "parentRef": {"name": "not-ours"},
            "conditions": [{"type": "Accepted", "status": "True"},
                           {"type": "ResolvedRefs", "status": "True"}]}]}}
        assert not is_http_route_ready(gw, other)
        assert not is_http_route_ready(gw, None)

    def test_step_held_until_route_accepted(self):
        """Without any gateway controller writing status, weights stay 0."""
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.kube.events import StoreRecorder
        from kuberay_amd.ops.incremental import IncrementalUpgrader
        client = InMemoryClient()
        svc = make_service()
        svc = client.create(svc)
        active = client.create(simple_raycluster("active-c", workers=1))
        pending = client.create(simple_raycluster("pending-c", workers=1))
        up = IncrementalUpgrader(client, StoreRecorder(client.server))
        up.ensure_gateway_infra(svc, active, pending)
        # no sim kubelet: Gateway/HTTPRoute status never set -> hold
        assert up.step_traffic(svc, active, pending) is False
        assert (svc.status.pending_service_status.traffic_routed_percent
                or 0) == 0
        # hand-program the gateway + route like a gateway controller would
        ns = "default"
        client.server.patch_merge(
            "Gateway", ns, f"{svc.metadata.name}-gateway",
            {"status": {"conditions": [
                {"type": "Accepted", "status": "True"},
                {"type": "Programmed", "status": "True"}]}},
            subresource="status")
        client.server.patch_merge(
            "HTTPRoute", ns, f"{svc.metadata.name}-route",
            {"status": {"parents": [{
                "parentRef": {"name": f"{svc.metadata.name}-gateway"},
                "conditions": [
                    {"type": "Accepted", "status": "True"},
                    {"type": "ResolvedRefs", "status": "True"}]}]}},
            subresource="status")
        assert up.step_traffic(svc, active, pending) is False  # 50%
        assert svc.status.pending_service_status.traffic_routed_percent == 50


class TestSteppingInvariants:
    def _ready_gateway(self, client, svc):
        ns = "default"
        client.server.patch_merge(
            "Gateway", ns, f"{svc.metadata.name}-gateway",
            {"status": {"conditions": [
                {"type": "Accepted", "status": "True"},
                {"type": "Programmed", "status": "True"}]}},
            subresource="status")
        client.server.patch_merge(
            "HTTPRoute", ns, f"{svc.metadata.name}-route",
            {"status": {"parents": [{
                "parentRef": {"name": f"{svc.metadata.name}-gateway"},
                "conditions": [
                    {"type": "Accepted", "status": "True"},
                    {"type": "ResolvedRefs", "status": "True"}]}]}},
            subresource="status")

    @pytest.mark.parametrize("step", [1, 7, 25, 40, 100])
    def test_weight_walk_invariants(self, step):
        """For ANY stepSizePercent: weights are monotone, bounded by the
        step, active+pending always sum to 100, capacity leads traffic,
        promotion fires exactly at 100, and the HTTPRoute carries the
        same split (rayservice_controller.go:1644-1843 invariants)."""
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.kube.events import StoreRecorder
        from kuberay_amd.ops.incremental import IncrementalUpgrader
        client = InMemoryClient()
        svc = client.create(make_service(interval=0, step=step))
        active = client.create(simple_raycluster("act", workers=1))
        pending = client.create(simple_raycluster("pen", workers=1))
        up = IncrementalUpgrader(client, StoreRecorder(client.server))
        up.ensure_gateway_infra(svc, active, pending)
        self._ready_gateway(client, svc)
        prev = 0
        for i in range(0, 120):
            done = up.step_traffic(svc, active, pending)
            ps = svc.status.pending_service_status
            As = svc.status.active_service_status
            w = ps.traffic_routed_percent or 0
            assert prev < w <= min(100, prev + step)
            assert (As.traffic_routed_percent or 0) + w == 100
            assert (ps.target_capacity or 0) >= w
            route = client.server.get(
                "HTTPRoute", "default", f"{svc.metadata.name}-route")
            weights = {r["name"]: r["weight"] for r in
                       route["spec"]["rules"][0]["backendRefs"]}
            assert sum(weights.values()) == 100
            assert done is (w >= 100)
            if done:
                break
            prev = w
        assert prev < 100 and w == 100
