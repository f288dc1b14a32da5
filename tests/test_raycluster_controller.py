"""RayCluster reconciler tests against the in-process control plane
(reference analog: raycluster_controller_test.go / _unit_test.go under
envtest; SURVEY.md §4 tier 2)."""
import time

import pytest

from kuberay_amd.kube import objects as k8s
from kuberay_amd.models import RayCluster
from kuberay_amd.ops.raycluster import should_delete_pod
from kuberay_amd.testing import simple_raycluster
from kuberay_amd.utils import constants as C


def get_cluster(cp, name="demo"):
    return cp.client.get(RayCluster, "default", name)


def pods_of(cp, name="demo"):
    return [p for p in cp.server.list("Pod")
            if (p["metadata"].get("labels") or {}).get(C.RAY_CLUSTER_LABEL_KEY) == name]


class TestLifecycle:
    def test_cluster_reaches_ready(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=2))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        rc = get_cluster(control_plane)
        assert rc.status.desired_worker_replicas == 2
        assert rc.status.available_worker_replicas == 2
        assert rc.status.ready_worker_replicas == 2
        assert rc.status.state_transition_times.get("ready")
        assert rc.status.observed_generation == rc.metadata.generation
        assert rc.status.head.pod_ip
        assert rc.status.endpoints["dashboard"] == "8265"

    def test_conditions(self, control_plane):
        control_plane.client.create(simple_raycluster("demo"))
        assert control_plane.wait_cluster_condition("default", "demo", "HeadPodReady")
        assert control_plane.wait_cluster_condition("default", "demo", "RayClusterProvisioned")

    def test_head_service_created(self, control_plane):
        control_plane.client.create(simple_raycluster("demo"))
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get("Service", "default", "demo-head-svc"))
        svc = control_plane.server.get("Service", "default", "demo-head-svc")
        assert svc["metadata"]["ownerReferences"][0]["kind"] == "RayCluster"

    def test_desired_gpu_status(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=2, gpus_per_worker=4))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        assert get_cluster(control_plane).status.desired_gpu == "8"

    def test_cluster_deletion_cascades_to_pods(self, control_plane):
        control_plane.client.create(simple_raycluster("demo"))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        control_plane.server.delete("RayCluster", "default", "demo")
        assert control_plane.wait_for(lambda: len(pods_of(control_plane)) == 0)


class TestScaling:
    def test_scale_up(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=1))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].replicas = 3
        control_plane.client.update(rc)
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.ready_worker_replicas == 3)

    def test_scale_down_without_autoscaler_deletes_random(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=3))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].replicas = 1
        control_plane.client.update(rc)
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 1)

    def test_workers_to_delete_honored(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=2))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        victim = next(p["metadata"]["name"] for p in pods_of(control_plane)
                      if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY] == "worker")
        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].replicas = 1
        rc.spec.worker_group_specs[0].scale_strategy.workers_to_delete = [victim]
        control_plane.client.update(rc)
        assert control_plane.wait_for(
            lambda: victim not in [p["metadata"]["name"] for p in pods_of(control_plane)])
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 1)

    def test_autoscaler_blocks_random_delete(self, control_plane):
        cluster = simple_raycluster("demo", workers=3, enableInTreeAutoscaling=True)
        control_plane.client.create(cluster)
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 3)
        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].replicas = 1
        control_plane.client.update(rc)
        time.sleep(0.5)
        # without WorkersToDelete, operator must NOT delete pods when autoscaling
        assert get_cluster(control_plane).status.available_worker_replicas == 3

    def test_group_suspend(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=2))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].suspend = True
        control_plane.client.update(rc)
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 0)
        assert get_cluster(control_plane).status.desired_worker_replicas == 0

    def test_multihost_group(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=2, num_of_hosts=2))
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 4)
        # headless service for multi-host groups
        assert control_plane.server.try_get("Service", "default", "demo-headless")


class TestSuspendResume:
    def test_suspend_deletes_all_pods_and_sets_conditions(self, control_plane):
        control_plane.client.create(simple_raycluster("demo"))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        rc = get_cluster(control_plane)
        rc.spec.suspend = True
        control_plane.client.update(rc)
        assert control_plane.wait_cluster_state("default", "demo", "suspended")
        assert len(pods_of(control_plane)) == 0
        assert control_plane.wait_cluster_condition("default", "demo", "RayClusterSuspended")

    def test_resume(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", suspend=True))
        assert control_plane.wait_cluster_state("default", "demo", "suspended")
        rc = get_cluster(control_plane)
        rc.spec.suspend = False
        control_plane.client.update(rc)
        assert control_plane.wait_cluster_state("default", "demo", "ready")


class TestFailureRecovery:
    def test_failed_worker_pod_recreated(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=1))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        worker = next(p for p in pods_of(control_plane)
                      if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY] == "worker")
        control_plane.server.patch_merge(
            "Pod", "default", worker["metadata"]["name"],
            {"status": {"phase": "Failed"}}, subresource="status")
        def recreated():
            workers = [p for p in pods_of(control_plane)
                       if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY] == "worker"]
            return (len(workers) == 1
                    and workers[0]["metadata"]["name"] != worker["metadata"]["name"]
                    and workers[0].get("status", {}).get("phase") == "Running")
        assert control_plane.wait_for(recreated)

    def test_failed_head_pod_recreated(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=0))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        head = next(p for p in pods_of(control_plane)
                    if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY] == "head")
        control_plane.server.patch_merge(
            "Pod", "default", head["metadata"]["name"],
            {"status": {"phase": "Failed"}}, subresource="status")
        def recreated():
            heads = [p for p in pods_of(control_plane)
                     if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY] == "head"]
            return (len(heads) == 1
                    and heads[0]["metadata"]["name"] != head["metadata"]["name"])
        assert control_plane.wait_for(recreated)


class TestShouldDeletePod:
    def _pod(self, phase, restart_policy="Always", terminated=False):
        pod = k8s.Pod.from_dict({
            "metadata": {"name": "p"},
            "spec": {"containers": [{"name": "ray"}], "restartPolicy": restart_policy},
            "status": {"phase": phase},
        })
        if terminated:
            pod.status.container_statuses = [k8s.ContainerStatus.from_dict(
                {"name": "ray", "state": {"terminated": {"exitCode": 1}}})]
        return pod

    def test_terminal_phases(self):
        assert should_delete_pod(self._pod("Failed"), "worker")[0]
        assert should_delete_pod(self._pod("Succeeded"), "worker")[0]

    def test_running_healthy(self):
        assert not should_delete_pod(self._pod("Running"), "worker")[0]

    def test_terminated_container_restart_never(self):
        pod = self._pod("Running", restart_policy="Never", terminated=True)
        assert should_delete_pod(pod, "worker")[0]

    def test_terminated_container_restart_always(self):
        pod = self._pod("Running", restart_policy="Always", terminated=True)
        assert not should_delete_pod(pod, "worker")[0]


class TestValidationRejection:
    def test_invalid_spec_sets_failed_state(self, control_plane):
        bad = simple_raycluster("demo")
        bad.spec.worker_group_specs[0].min_replicas = 5
        bad.spec.worker_group_specs[0].max_replicas = 1
        control_plane.client.create(bad)
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.state == "failed")
        assert "minReplicas" in get_cluster(control_plane).status.reason


class TestGcsFaultTolerance:
    @pytest.fixture(autouse=True)
    def _embedded_gate(self):
        import kuberay_amd.features as features
        features.set_gate("GCSFaultToleranceEmbeddedStorage", True)
        yield
        features.reset()

    def test_redis_cleanup_finalizer_machine(self, control_plane):
        cluster = simple_raycluster("demo", gcsFaultToleranceOptions={
            "backend": "redis", "redisAddress": "redis://r:6379"})
        control_plane.client.create(cluster)
        assert control_plane.wait_for(
            lambda: C.GCS_FT_REDIS_CLEANUP_FINALIZER in
            (get_cluster(control_plane).metadata.finalizers or []))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        control_plane.server.delete("RayCluster", "default", "demo")
        # cleanup job created, completes (sim kubelet), finalizer removed, CR gone
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get("RayCluster", "default", "demo") is None,
            timeout=15)

    def test_embedded_backend_creates_pvc(self, control_plane):
        cluster = simple_raycluster("demo", gcsFaultToleranceOptions={
            "backend": "embedded", "storage": {"size": "2Gi"}})
        control_plane.client.create(cluster)
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get(
                "PersistentVolumeClaim", "default", "demo-gcs-pvc"))


class TestAutoscalerRBAC:
    def test_rbac_objects_created(self, control_plane):
        control_plane.client.create(
            simple_raycluster("demo", enableInTreeAutoscaling=True))
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get("ServiceAccount", "default", "demo"))
        role = control_plane.server.get("Role", "default", "demo")
        resources = {r for rule in role["rules"] for r in rule["resources"]}
        assert "rayclusters" in resources and "pods" in resources


class TestRandomPodDeleteFlag:
    def test_enable_random_pod_delete_overrides_autoscaler_gate(
            self, control_plane, monkeypatch):
        monkeypatch.setenv(C.ENABLE_RANDOM_POD_DELETE, "true")
        control_plane.client.create(
            simple_raycluster("demo", workers=3, enableInTreeAutoscaling=True))
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 3)
        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].replicas = 1
        control_plane.client.update(rc)
        # with the flag on, the operator random-deletes down to 1 even
        # though autoscaling is enabled (reference ENABLE_RANDOM_POD_DELETE)
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 1)


class TestWorkerGroupAddition:
    def test_new_group_added_to_live_cluster(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=1))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        rc = get_cluster(control_plane)
        from kuberay_amd.models.raycluster import WorkerGroupSpec
        new_group = {
            "groupName": "gpu-group", "replicas": 2, "minReplicas": 0,
            "maxReplicas": 4, "rayStartParams": {},
            "template": {"spec": {"containers": [{
                "name": "ray-worker", "image": "rayproject/ray:2.46.0-rocm",
                "resources": {"limits": {"cpu": "1", "memory": "1Gi",
                                         "amd.com/gpu": "1"}}}]}},
        }
        rc.spec.worker_group_specs.append(WorkerGroupSpec.from_dict(new_group))
        control_plane.client.update(rc)
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 3)
        group_pods = [p for p in pods_of(control_plane)
                      if (p["metadata"]["labels"] or {}).get(
                          C.RAY_NODE_GROUP_LABEL_KEY) == "gpu-group"]
        assert len(group_pods) == 2
        assert get_cluster(control_plane).status.desired_gpu == "2"


class TestGcsPvcRetain:
    @pytest.fixture(autouse=True)
    def _embedded_gate(self):
        import kuberay_amd.features as features
        features.set_gate("GCSFaultToleranceEmbeddedStorage", True)
        yield
        features.reset()

    def test_retain_policy_pvc_survives_cluster_deletion(self, control_plane):
        cluster = simple_raycluster("demo", gcsFaultToleranceOptions={
            "backend": "embedded",
            "storage": {"size": "1Gi", "deletionPolicy": "Retain"}})
        control_plane.client.create(cluster)
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get(
                "PersistentVolumeClaim", "default", "demo-gcs-pvc"))
        control_plane.server.delete("RayCluster", "default", "demo")
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get("RayCluster", "default", "demo") is None)
        # Retain: the PVC is NOT owner-referenced, so it survives GC
        assert control_plane.server.try_get(
            "PersistentVolumeClaim", "default", "demo-gcs-pvc") is not None
        # ADOPTION: recreating the cluster under the same name reuses the
        # retained PVC — the GCS state survives the recreate
        # (raycluster_controller.go:657-743 retain/adopt semantics)
        pvc_uid = control_plane.server.get(
            "PersistentVolumeClaim", "default", "demo-gcs-pvc"
        )["metadata"]["uid"]
        control_plane.client.create(simple_raycluster(
            "demo", gcsFaultToleranceOptions={
                "backend": "embedded",
                "storage": {"size": "1Gi", "deletionPolicy": "Retain"}}))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        assert control_plane.server.get(
            "PersistentVolumeClaim", "default", "demo-gcs-pvc"
        )["metadata"]["uid"] == pvc_uid, "retained PVC was replaced"

    def test_delete_policy_pvc_garbage_collected(self, control_plane):
        cluster = simple_raycluster("demo", gcsFaultToleranceOptions={
            "backend": "embedded",
            "storage": {"size": "1Gi", "deletionPolicy": "Delete"}})
        control_plane.client.create(cluster)
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get(
                "PersistentVolumeClaim", "default", "demo-gcs-pvc"))
        control_plane.server.delete("RayCluster", "default", "demo")
        assert control_plane.wait_for(
            lambda: control_plane.server.try_get(
                "PersistentVolumeClaim", "default", "demo-gcs-pvc") is None)


class TestMultihostAtomicity:
    def test_replica_units_share_group_label(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=2,
                                                      num_of_hosts=3))
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 6)
        views = control_plane.client.list_pod_views(
            "default", {C.RAY_CLUSTER_LABEL_KEY: "demo",
                        C.RAY_NODE_TYPE_LABEL_KEY: "worker"})
        by_rep = {}
        for v in views:
            rep = v.labels[C.RAY_WORKER_REPLICA_NAME_KEY]
            by_rep.setdefault(rep, []).append(v)
        assert len(by_rep) == 2
        for rep, members in by_rep.items():
            assert len(members) == 3
            hosts = sorted(v.labels[C.RAY_HOST_INDEX_KEY] for v in members)
            assert hosts == ["0", "1", "2"]

    def test_partial_replica_rebuilt_whole(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=1,
                                                      num_of_hosts=3))
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 3)
        views = control_plane.client.list_pod_views(
            "default", {C.RAY_CLUSTER_LABEL_KEY: "demo",
                        C.RAY_NODE_TYPE_LABEL_KEY: "worker"})
        old_rep = views[0].labels[C.RAY_WORKER_REPLICA_NAME_KEY]
        victim = views[0].name
        # kill one host of the replica
        control_plane.server.patch_merge("Pod", "default", victim,
                                         {"status": {"phase": "Failed"}},
                                         subresource="status")
        def rebuilt():
            vs = control_plane.client.list_pod_views(
                "default", {C.RAY_CLUSTER_LABEL_KEY: "demo",
                            C.RAY_NODE_TYPE_LABEL_KEY: "worker"})
            live = [v for v in vs if not v.deletion_timestamp
                    and v.phase == "Running"]
            if len(live) != 3:
                return False
            reps = {v.labels[C.RAY_WORKER_REPLICA_NAME_KEY] for v in live}
            # the whole unit was replaced under a fresh replica name
            return len(reps) == 1 and old_rep not in reps
        assert control_plane.wait_for(rebuilt, timeout=20)


class TestUpgradeStrategyRecreate:
    def test_pods_rebuilt_on_spec_change(self, control_plane):
        cluster = simple_raycluster("demo", workers=2,
                                    upgradeStrategy={"type": "Recreate"})
        control_plane.client.create(cluster)
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        old_names = {p["metadata"]["name"] for p in pods_of(control_plane)}

        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].template.spec.containers[0].image = \
            "rayproject/ray:2.47.0-rocm"
        control_plane.client.update(rc)

        def rebuilt():
            current = {p["metadata"]["name"] for p in pods_of(control_plane)}
            rc2 = get_cluster(control_plane)
            return (current.isdisjoint(old_names) and len(current) == 3
                    and rc2.status.state == "ready"
                    and rc2.status.ready_worker_replicas == 2)
        assert control_plane.wait_for(rebuilt, timeout=20)

    def test_no_recreate_without_strategy(self, control_plane):
        control_plane.client.create(simple_raycluster("demo", workers=1))
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        old_names = {p["metadata"]["name"] for p in pods_of(control_plane)}
        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].template.spec.containers[0].image = \
            "rayproject/ray:2.47.0-rocm"
        control_plane.client.update(rc)
        time.sleep(0.6)
        current = {p["metadata"]["name"] for p in pods_of(control_plane)}
        assert current == old_names  # default: in-place, no recreation

    def test_replicas_change_does_not_recreate(self, control_plane):
        cluster = simple_raycluster("demo", workers=1,
                                    upgradeStrategy={"type": "Recreate"})
        control_plane.client.create(cluster)
        assert control_plane.wait_cluster_state("default", "demo", "ready")
        old_names = {p["metadata"]["name"] for p in pods_of(control_plane)}
        rc = get_cluster(control_plane)
        rc.spec.worker_group_specs[0].replicas = 2
        control_plane.client.update(rc)
        assert control_plane.wait_for(
            lambda: get_cluster(control_plane).status.available_worker_replicas == 2)
        current = {p["metadata"]["name"] for p in pods_of(control_plane)}
        assert old_names.issubset(current)  # hash mutes replica changes


class TestGcsFtHeadRecovery:
    """e2e/gcs_ft e2e analog: with GCS FT enabled, a head crash is
    recovered by RECREATING ONLY THE HEAD — worker pods survive untouched
    and reconnect to the restored GCS (the reference relies on the 600 s
    RAY_gcs_rpc_server_reconnect_timeout_s it injects)."""

    def test_head_crash_keeps_workers_alive(self, control_plane):
        cp = control_plane
        cp.client.create(simple_raycluster(
            "ftdemo", workers=2,
            gcsFaultToleranceOptions={"backend": "redis",
                                      "redisAddress": "redis://r:6379"}))
        assert cp.wait_cluster_state("default", "ftdemo", "ready")
        pods = cp.server.list("Pod", "default", {"ray.io/cluster": "ftdemo"})
        workers_before = sorted(
            p["metadata"]["name"] for p in pods
            if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY] == "worker")
        head = next(p for p in pods
                    if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY]
                    == "head")
        assert len(workers_before) == 2
        # head crashes
        cp.server.patch_merge("Pod", "default", head["metadata"]["name"],
                              {"status": {"phase": "Failed"}},
                              subresource="status")

        def head_recreated():
            heads = [p for p in cp.server.list(
                "Pod", "default", {"ray.io/cluster": "ftdemo"})
                if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY]
                == "head"]
            return (len(heads) == 1 and heads[0]["metadata"]["name"]
                    != head["metadata"]["name"]
                    and heads[0].get("status", {}).get("phase") == "Running")
        assert cp.wait_for(head_recreated)
        assert cp.wait_cluster_state("default", "ftdemo", "ready")
        # the original workers were never touched
        workers_after = sorted(
            p["metadata"]["name"] for p in cp.server.list(
                "Pod", "default", {"ray.io/cluster": "ftdemo"})
            if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY]
            == "worker")
        assert workers_after == workers_before

    def test_ft_env_reaches_worker_pods(self, control_plane):
        """The reconnect-timeout env the recovery story depends on is
        actually injected into worker pods (pod.go:94-102 analog)."""
        cp = control_plane
        cp.client.create(simple_raycluster(
            "ftenv", workers=1,
            gcsFaultToleranceOptions={"backend": "redis",
                                      "redisAddress": "redis://r:6379"}))
        assert cp.wait_cluster_state("default", "ftenv", "ready")
        worker = next(p for p in cp.server.list(
            "Pod", "default", {"ray.io/cluster": "ftenv"})
            if p["metadata"]["labels"][C.RAY_NODE_TYPE_LABEL_KEY]
            == "worker")
        env = {e["name"]: e.get("value") for e in
               worker["spec"]["containers"][0].get("env", [])}
        assert "RAY_gcs_rpc_server_reconnect_timeout_s" in env


class TestIngressReconcile:
    """reconcileIngress e2e (raycluster_controller.go:496-606): the head
    Ingress appears when enableIngress is set and is owned by the cluster."""

    def test_ingress_created_when_enabled(self, control_plane):
        cp = control_plane
        rc = simple_raycluster("ingdemo", workers=0)
        rc.spec.head_group_spec.enable_ingress = True
        cp.client.create(rc)
        assert cp.wait_cluster_state("default", "ingdemo", "ready")
        assert cp.wait_for(
            lambda: cp.server.count("Ingress") == 1, timeout=10)
        ing = cp.server.list("Ingress")[0]
        owner = ing["metadata"]["ownerReferences"][0]
        assert owner["kind"] == "RayCluster" and owner["name"] == "ingdemo"
        # routes to the head service's dashboard port
        rule = ing["spec"]["rules"][0]
        backend = rule["http"]["paths"][0]["backend"]["service"]
        assert backend["name"] == "ingdemo-head-svc"

    def test_no_ingress_by_default(self, control_plane):
        cp = control_plane
        cp.client.create(simple_raycluster("noing", workers=0))
        assert cp.wait_cluster_state("default", "noing", "ready")
        assert cp.server.count("Ingress") == 0


class TestReplicaRandomWalk:
    def test_replica_random_walk_converges_exactly(self, control_plane):
        """Random replica walk (0..5) with interleaved waits: after the
        final target the cluster must converge to EXACTLY that many ready
        workers — no stragglers from superseded targets, no over-delete."""
        import random
        rng = random.Random(13)
        from kuberay_amd.models import RayCluster
        control_plane.client.create(simple_raycluster("walk", workers=1))
        assert control_plane.wait_cluster_state("default", "walk", "ready")
        target = 1
        for _ in range(8):
            target = rng.randint(0, 5)
            control_plane.client.update_with_retry(
                RayCluster, "default", "walk",
                lambda c, t=target: (
                    setattr(c.spec.worker_group_specs[0], "replicas", t),
                    setattr(c.spec.worker_group_specs[0], "max_replicas",
                            max(t, c.spec.worker_group_specs[0]
                                .max_replicas or 0))))
            time.sleep(rng.uniform(0.02, 0.25))
        assert control_plane.wait_for(
            lambda: control_plane.client.get(
                RayCluster, "default", "walk"
            ).status.ready_worker_replicas == target, timeout=40), target
        # pod-level exactness, not just status
        def worker_pods():
            return [v for v in control_plane.client.list_pod_views("default")
                    if v.labels.get("ray.io/node-type") == "worker"
                    and v.labels.get("ray.io/cluster") == "walk"
                    and not v.deletion_timestamp]
        assert control_plane.wait_for(
            lambda: len(worker_pods()) == target, timeout=30), \
            (target, len(worker_pods()))
