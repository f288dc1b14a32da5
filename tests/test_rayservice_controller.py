"""RayService controller tests (reference analog: rayservice_controller_test.go)."""
import time

import pytest

from kuberay_amd.models import RayCluster, RayService
from kuberay_amd.models.rayservice import ApplicationStatus
from kuberay_amd.testing import simple_raycluster
from kuberay_amd.utils import constants as C

SERVE_CONFIG = """\
applications:
- name: app1
  import_path: mod.graph
  deployments:
  - name: D1
  - name: D2
"""


def make_rayservice(name="svc1", **spec_overrides):
    spec = {
        "serveConfigV2": SERVE_CONFIG,
        "rayClusterConfig": simple_raycluster("x", workers=1).spec.to_dict(),
    }
    spec.update(spec_overrides)
    return RayService.from_dict({
        "apiVersion": "ray.io/v1", "kind": "RayService",
        "metadata": {"name": name, "namespace": "default"},
        "spec": spec,
    })


def update_svc_with_retry(cp, mutate, attempts=20):
    """get-mutate-update with optimistic-concurrency retry: the reconciler
    can bump resourceVersion between our get and update."""
    from kuberay_amd.kube.store import ConflictError
    for _ in range(attempts):
        svc = svc_of(cp)
        mutate(svc)
        try:
            cp.client.update(svc)
            return
        except ConflictError:
            time.sleep(0.05)
    raise AssertionError("update kept conflicting")


def svc_of(cp, name="svc1"):
    return cp.client.get(RayService, "default", name)


def wait_ready(cp, name="svc1", timeout=20):
    return cp.wait_for(lambda: svc_of(cp, name).condition_true("Ready"), timeout)


@pytest.fixture()
def fast_gc(control_plane):
    control_plane.rayservice_reconciler.cluster_deletion_delay_s = 0.2
    return control_plane


class TestDeploy:
    def test_service_becomes_ready(self, control_plane):
        control_plane.client.create(make_rayservice())
        assert wait_ready(control_plane)
        svc = svc_of(control_plane)
        assert svc.status.service_status == "Running"
        assert svc.status.num_serve_endpoints > 0
        active = svc.status.active_service_status
        assert active.ray_cluster_name
        assert active.applications["app1"].status == ApplicationStatus.RUNNING
        assert set(active.applications["app1"].deployments) == {"D1", "D2"}

    def test_services_created_and_point_to_active(self, control_plane):
        control_plane.client.create(make_rayservice())
        assert wait_ready(control_plane)
        active = svc_of(control_plane).status.active_service_status.ray_cluster_name
        head = control_plane.server.get("Service", "default", "svc1-head-svc")
        serve = control_plane.server.get("Service", "default", "svc1-serve-svc")
        assert head["spec"]["selector"][C.RAY_CLUSTER_LABEL_KEY] == active
        assert serve["spec"]["selector"][C.RAY_CLUSTER_LABEL_KEY] == active
        assert serve["spec"]["selector"][C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY] == "true"

    def test_serve_config_submitted_once_per_config(self, control_plane):
        control_plane.client.create(make_rayservice())
        assert wait_ready(control_plane)
        time.sleep(0.3)  # several reconcile cycles
        assert len(control_plane.dashboard.update_serve_calls) == 1

    def test_head_pod_serve_label_flipped(self, control_plane):
        control_plane.client.create(make_rayservice())
        assert wait_ready(control_plane)
        active = svc_of(control_plane).status.active_service_status.ray_cluster_name
        def head_labeled():
            pods = control_plane.server.list(
                "Pod", "default", {C.RAY_CLUSTER_LABEL_KEY: active,
                                   C.RAY_NODE_TYPE_LABEL_KEY: "head"})
            return pods and pods[0]["metadata"]["labels"].get(
                C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY) == "true"
        assert control_plane.wait_for(head_labeled)


class TestZeroDowntimeUpgrade:
    def test_spec_change_promotes_new_cluster(self, fast_gc):
        cp = fast_gc
        cp.client.create(make_rayservice())
        assert wait_ready(cp)
        old_active = svc_of(cp).status.active_service_status.ray_cluster_name

        def set_image(svc):
            svc.spec.ray_cluster_spec.worker_group_specs[0].template.spec \
                .containers[0].image = "rayproject/ray:2.47.0-rocm"
        update_svc_with_retry(cp, set_image)

        def promoted():
            s = svc_of(cp)
            return (s.status.active_service_status.ray_cluster_name
                    not in (None, old_active) and s.condition_true("Ready"))
        assert cp.wait_for(promoted, timeout=25)
        # old cluster eventually GC'd after the deletion delay
        assert cp.wait_for(
            lambda: cp.server.try_get("RayCluster", "default", old_active) is None,
            timeout=15)

    def test_replicas_only_change_updates_in_place(self, control_plane):
        cp = control_plane
        cp.client.create(make_rayservice())
        assert wait_ready(cp)
        active = svc_of(cp).status.active_service_status.ray_cluster_name
        def set_replicas(svc):
            svc.spec.ray_cluster_spec.worker_group_specs[0].replicas = 3
        update_svc_with_retry(cp, set_replicas)
        def scaled():
            rc = cp.client.try_get(RayCluster, "default", active)
            return rc is not None and rc.status.available_worker_replicas == 3
        assert cp.wait_for(scaled, timeout=20)
        # no new cluster was prepared
        assert svc_of(cp).status.active_service_status.ray_cluster_name == active

    def test_upgrade_strategy_none_blocks_new_cluster(self, control_plane):
        cp = control_plane
        cp.client.create(make_rayservice(upgradeStrategy={"type": "None"}))
        assert wait_ready(cp)
        active = svc_of(cp).status.active_service_status.ray_cluster_name
        def set_image(svc):
            svc.spec.ray_cluster_spec.worker_group_specs[0].template.spec \
                .containers[0].image = "rayproject/ray:2.47.0-rocm"
        update_svc_with_retry(cp, set_image)
        time.sleep(0.5)
        s = svc_of(cp)
        assert s.status.active_service_status.ray_cluster_name == active
        assert not s.status.pending_service_status.ray_cluster_name


class TestOrphanClusterSweep:
    def test_replaced_cluster_survives_operator_restart(self):
        """Operator restart inside the deletion-delay window must not leak
        the replaced active cluster: the orphan sweep rediscovers it from the
        API server (reference reconcileRayCluster GC of non-active/pending
        owned clusters)."""
        from kuberay_amd.testing import ControlPlane
        cp1 = ControlPlane(kubelet_delay=0.01, job_runtime=0.2,
                           poll_seconds=0.05)
        # long delay: the old cluster is still pending deletion at "crash"
        cp1.rayservice_reconciler.cluster_deletion_delay_s = 300
        cp1.start()
        try:
            cp1.client.create(make_rayservice())
            assert wait_ready(cp1)
            old_active = svc_of(cp1).status.active_service_status.ray_cluster_name

            def set_image(svc):
                svc.spec.ray_cluster_spec.worker_group_specs[0].template.spec \
                    .containers[0].image = "rayproject/ray:2.47.0-rocm"
            update_svc_with_retry(cp1, set_image)

            def promoted():
                s = svc_of(cp1)
                return (s.status.active_service_status.ray_cluster_name
                        not in (None, old_active) and s.condition_true("Ready"))
            assert cp1.wait_for(promoted, timeout=25)
            # old cluster still exists — deletion is pending in cp1's memory
            assert cp1.server.try_get("RayCluster", "default", old_active)
        finally:
            cp1.stop()  # operator "crash" loses the in-memory timer

        cp2 = ControlPlane(kubelet_delay=0.01, job_runtime=0.2,
                           poll_seconds=0.05, server=cp1.server)
        cp2.rayservice_reconciler.cluster_deletion_delay_s = 0.2
        cp2.start()
        try:
            assert cp2.wait_for(
                lambda: cp2.server.try_get("RayCluster", "default",
                                           old_active) is None,
                timeout=15), "replaced cluster leaked across operator restart"
            # the promoted active cluster is untouched
            s = svc_of(cp2)
            assert cp2.server.try_get(
                "RayCluster", "default",
                s.status.active_service_status.ray_cluster_name)
        finally:
            cp2.stop()


class TestSuspend:
    def test_suspend_and_resume(self, control_plane):
        cp = control_plane
        cp.client.create(make_rayservice())
        assert wait_ready(cp)
        def suspend_on(svc):
            svc.spec.suspend = True
        update_svc_with_retry(cp, suspend_on)
        assert cp.wait_for(lambda: svc_of(cp).condition_true("Suspended"), timeout=15)
        assert cp.wait_for(lambda: cp.server.count("RayCluster") == 0)
        assert svc_of(cp).status.num_serve_endpoints == 0

        def suspend_off(svc):
            svc.spec.suspend = False
        update_svc_with_retry(cp, suspend_off)
        assert cp.wait_for(lambda: wait_ready(cp), timeout=25)


class TestUnhealthy:
    def test_deploy_failed_app_keeps_service_not_ready(self, control_plane):
        cp = control_plane
        cp.dashboard.serve_statuses_mock = {"applications": {
            "app1": {"status": "DEPLOY_FAILED", "message": "import error",
                     "deployments": {}}}}
        cp.client.create(make_rayservice())
        time.sleep(1.0)
        s = svc_of(cp)
        # first cluster stays pending; service never flips Ready
        assert not s.condition_true("Ready")
        apps = s.status.pending_service_status.applications or {}
        assert apps.get("app1") and apps["app1"].status == "DEPLOY_FAILED"


class TestServeStatusEdgeCases:
    def test_app_vanishes_from_dashboard_flips_ready_off(self, control_plane):
        cp = control_plane
        cp.client.create(make_rayservice())
        assert wait_ready(cp)
        # dashboard suddenly reports no applications (e.g. serve controller died)
        cp.dashboard.serve_statuses_mock = {"applications": {}}
        def not_ready():
            s = svc_of(cp)
            return (not s.condition_true("Ready")
                    or s.status.num_serve_endpoints == 0)
        # Ready condition tracks endpoints; pods are still serving so the
        # endpoints stay — but application status must reflect the outage
        def apps_gone():
            s = svc_of(cp)
            apps = s.status.active_service_status.applications
            return apps == {} or apps is None or all(
                a.status != "RUNNING" for a in apps.values())
        assert cp.wait_for(apps_gone, timeout=15)

    def test_dashboard_flap_keeps_config_cached(self, control_plane):
        """A transient status-poll failure must not resubmit the serve
        config (only submission failures invalidate the cache; the cache is
        keyed by cluster UID so recreation still resubmits — see
        tests/test_rayservice_adversarial.py)."""
        cp = control_plane
        cp.client.create(make_rayservice())
        assert wait_ready(cp)
        calls_before = len(cp.dashboard.update_serve_calls)
        from kuberay_amd.utils.dashboard_client import DashboardClientError
        original = cp.dashboard.get_serve_applications
        state = {"fail": 2}
        def flaky():
            if state["fail"] > 0:
                state["fail"] -= 1
                raise DashboardClientError("dashboard restarting")
            return original()
        cp.dashboard.get_serve_applications = flaky
        assert cp.wait_for(lambda: state["fail"] == 0, timeout=15)
        assert cp.wait_for(lambda: svc_of(cp).condition_true("Ready"),
                           timeout=15)
        assert len(cp.dashboard.update_serve_calls) == calls_before


class TestUnhealthyReplacement:
    def test_persistently_unhealthy_apps_trigger_replacement(self, control_plane):
        cp = control_plane
        cp.rayservice_reconciler.cluster_deletion_delay_s = 0.2
        cp.client.create(make_rayservice(serviceUnhealthySecondThreshold=1))
        assert wait_ready(cp)
        first_active = svc_of(cp).status.active_service_status.ray_cluster_name
        # the app goes permanently unhealthy on the active cluster
        cp.dashboard.serve_statuses_mock = {"applications": {
            "app1": {"status": "UNHEALTHY", "message": "actor died",
                     "deployments": {}}}}

        def replacement_started():
            s = svc_of(cp)
            return bool(s.status.pending_service_status.ray_cluster_name)
        assert cp.wait_for(replacement_started, timeout=20)
        # once the replacement cluster's apps are healthy, it promotes
        cp.dashboard.serve_statuses_mock = None

        def promoted():
            s = svc_of(cp)
            return (s.status.active_service_status.ray_cluster_name
                    not in (None, first_active) and s.condition_true("Ready"))
        assert cp.wait_for(promoted, timeout=25)

    def test_transient_unhealth_does_not_replace(self, control_plane):
        cp = control_plane
        cp.client.create(make_rayservice(serviceUnhealthySecondThreshold=3600))
        assert wait_ready(cp)
        cp.dashboard.serve_statuses_mock = {"applications": {
            "app1": {"status": "UNHEALTHY", "message": "blip",
                     "deployments": {}}}}
        time.sleep(0.8)
        cp.dashboard.serve_statuses_mock = None
        assert cp.wait_for(lambda: svc_of(cp).condition_true("Ready"),
                           timeout=15)
        assert not svc_of(cp).status.pending_service_status.ray_cluster_name
