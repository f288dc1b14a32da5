"""Operator wiring, feature gates, metrics, networkpolicy, gang scheduling."""
import time

import pytest

import kuberay_amd.features as features
from kuberay_amd.config import Configuration, load_config
from kuberay_amd.kube.client import InMemoryClient
from kuberay_amd.metrics import OperatorMetrics
from kuberay_amd.models import RayCluster
from kuberay_amd.operator import build_manager
from kuberay_amd.ops.networkpolicy import (
    NetworkPolicyReconciler,
    build_head_network_policy,
    build_worker_network_policies,
)
from kuberay_amd.parallel import (
    VolcanoBatchScheduler,
    XgmiGangScheduler,
    YunikornBatchScheduler,
    scheduler_for,
)
from kuberay_amd.testing import simple_raycluster
from kuberay_amd.utils import constants as C


@pytest.fixture(autouse=True)
def reset_gates():
    yield
    features.reset()


class TestFeatureGates:
    def test_defaults(self):
        assert features.enabled("RayClusterStatusConditions")
        assert not features.enabled("RayClusterMTLS")
        assert features.enabled("MI355XGpuHealthProbes")

    def test_parse(self):
        features.parse_feature_gates("RayClusterMTLS=true,RayCronJob=false")
        assert features.enabled("RayClusterMTLS")
        assert not features.enabled("RayCronJob")

    def test_unknown_gate_rejected(self):
        with pytest.raises(KeyError):
            features.set_gate("NoSuchGate", True)


class TestConfig:
    def test_defaults(self):
        cfg = load_config([])
        assert cfg.backend == "memory"
        assert cfg.reconcile_concurrency == 4

    def test_flags_override_file(self, tmp_path):
        f = tmp_path / "cfg.yaml"
        f.write_text("reconcileConcurrency: 2\nbatchScheduler: volcano\n")
        cfg = load_config(["--config", str(f), "--reconcile-concurrency", "8"])
        assert cfg.reconcile_concurrency == 8
        assert cfg.batch_scheduler == "volcano"


class TestOperatorWiring:
    def test_build_manager_controllers(self):
        cfg = Configuration()
        m, client, metrics, autoscaler = build_manager(cfg)
        names_ = [c.name for c in m.controllers]
        # RayCronJob is alpha/off by default (features.go:110)
        assert names_ == ["raycluster", "rayjob", "rayservice"]
        assert metrics is not None and autoscaler is not None

    def test_raycronjob_controller_gated(self):
        features.set_gate("RayCronJob", True)
        m, *_ = build_manager(Configuration())
        assert "raycronjob" in [c.name for c in m.controllers]

    def test_networkpolicy_controller_gated(self):
        features.set_gate("RayClusterNetworkPolicy", True)
        m, *_ = build_manager(Configuration())
        assert "networkpolicy" in [c.name for c in m.controllers]

    def test_end_to_end_via_operator_manager(self):
        cfg = Configuration(enable_mi355x_autoscaler=False)
        m, client, metrics, _ = build_manager(cfg)
        from kuberay_amd.kube.kubelet import SimKubelet
        kubelet = SimKubelet(m.server)
        m.start(); kubelet.start()
        try:
            client.create(simple_raycluster("op-e2e"))
            deadline = time.monotonic() + 15
            state = None
            while time.monotonic() < deadline:
                rc = client.try_get(RayCluster, "default", "op-e2e")
                state = rc.status.state if rc else None
                if state == "ready":
                    break
                time.sleep(0.05)
            assert state == "ready"
            # metrics recorded the provisioned cluster
            expo = metrics.exposition().decode()
            assert "kuberay_cluster_provisioned_duration_seconds" in expo
        finally:
            kubelet.stop(); m.stop()


class TestMetrics:
    def test_exposition_names(self):
        m = OperatorMetrics()
        expo = m.exposition().decode()
        for name in ("kuberay_cluster_provisioned_duration_seconds",
                     "kuberay_job_execution_duration_seconds",
                     "kuberay_service_info",
                     "kuberay_mi355x_gpu_utilization_pct"):
            assert name in expo, name

    def test_gpu_stats_observation(self):
        from kuberay_amd.gpu.rocm_smi import GpuStats
        m = OperatorMetrics()
        m.observe_gpu_stats([GpuStats(index=0, utilization_pct=50,
                                      vram_used_bytes=144 * 1024**3)])
        expo = m.exposition().decode()
        assert 'kuberay_mi355x_gpu_utilization_pct{gpu="0"} 50.0' in expo


class TestNetworkPolicy:
    def _cluster(self, mode="DenyAll", **np_extra):
        return simple_raycluster("demo", networkPolicy={"mode": mode, **np_extra})

    def test_head_policy_deny_all(self):
        p = build_head_network_policy(self._cluster())
        assert set(p.spec["policyTypes"]) == {"Ingress", "Egress"}
        # intra-cluster always allowed
        assert p.spec["ingress"][0]["from"][0]["podSelector"]["matchLabels"] == \
            {C.RAY_CLUSTER_LABEL_KEY: "demo"}
        # DNS egress open
        assert any(any(pp.get("port") == 53 for pp in r.get("ports", []))
                   for r in p.spec["egress"])

    def test_ingress_only_mode(self):
        p = build_head_network_policy(self._cluster(mode="DenyAllIngress"))
        assert p.spec["policyTypes"] == ["Ingress"]
        assert "egress" not in p.spec

    def test_worker_group_override(self):
        c = self._cluster(workerGroups=[{
            "groupName": "default-group",
            "ingressRules": [{"from": [{"ipBlock": {"cidr": "10.0.0.0/8"}}]}]}])
        policies = build_worker_network_policies(c)
        assert len(policies) == 2
        override = policies[1]
        assert override.spec["podSelector"]["matchLabels"][
            C.RAY_NODE_GROUP_LABEL_KEY] == "default-group"

    def test_reconciler_creates_and_gcs(self):
        client = InMemoryClient()
        cluster = self._cluster()
        client.create(cluster)
        r = NetworkPolicyReconciler(client)
        r.reconcile(("default", "demo"))
        assert client.server.count("NetworkPolicy") == 2
        # turn off policy -> stale GC
        rc = client.get(RayCluster, "default", "demo")
        rc.spec.network_policy = None
        client.update(rc)
        r.reconcile(("default", "demo"))
        assert client.server.count("NetworkPolicy") == 0


class TestGangScheduling:
    def _pod(self, cluster, gpus=1):
        from kuberay_amd.common import pod as podlib
        group = cluster.spec.worker_group_specs[0]
        fqdn = "x.default.svc.cluster.local"
        t = podlib.default_worker_pod_template(cluster, group, "p-", fqdn, "6379")
        return podlib.build_pod(t, "worker", group.ray_start_params, "6379",
                                False, None, fqdn)

    def test_selector_registry(self):
        assert isinstance(scheduler_for("volcano"), VolcanoBatchScheduler)
        assert isinstance(scheduler_for("xgmi-gang"), XgmiGangScheduler)
        assert scheduler_for(None) is None
        with pytest.raises(ValueError):
            scheduler_for("nope")

    def test_volcano_podgroup_and_annotations(self):
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=3, gpus_per_worker=1)
        client.create(cluster)
        sched = VolcanoBatchScheduler()
        sched.do_batch_scheduling_on_submission(client, cluster)
        pgs = client.server.list("PodGroup")
        assert len(pgs) == 1
        assert pgs[0]["spec"]["minMember"] == 4  # head + 3 workers
        assert pgs[0]["spec"]["minResources"]["amd.com/gpu"] == "3"
        pod = self._pod(cluster)
        sched.add_metadata_to_pod(client, cluster, "default-group", pod)
        assert pod.spec.scheduler_name == "volcano"
        assert pod.metadata.annotations["scheduling.k8s.io/group-name"] == \
            "ray-demo-pg"

    def test_xgmi_gang_adds_island_affinity_for_gpu_pods(self):
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=2, gpus_per_worker=4)
        client.create(cluster)
        sched = XgmiGangScheduler()
        sched.do_batch_scheduling_on_submission(client, cluster)
        assert client.server.count("PodGroup") == 1
        pod = self._pod(cluster)
        sched.add_metadata_to_pod(client, cluster, "default-group", pod)
        terms = pod.spec.affinity["podAffinity"][
            "preferredDuringSchedulingIgnoredDuringExecution"]
        assert terms[0]["podAffinityTerm"]["topologyKey"] == "amd.com/xgmi-island"
        assert pod.metadata.labels["ray.io/xgmi-gang"] == "demo-default-group"

    def test_xgmi_gang_skips_cpu_pods(self):
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=2, gpus_per_worker=0)
        client.create(cluster)
        sched = XgmiGangScheduler()
        sched.do_batch_scheduling_on_submission(client, cluster)
        assert client.server.count("PodGroup") == 0
        pod = self._pod(cluster, gpus=0)
        sched.add_metadata_to_pod(client, cluster, "default-group", pod)
        assert pod.spec.affinity is None

    def test_yunikorn_labels(self):
        client = InMemoryClient()
        cluster = simple_raycluster("demo")
        pod = self._pod(cluster)
        YunikornBatchScheduler().add_metadata_to_pod(client, cluster,
                                                     "default-group", pod)
        assert pod.metadata.labels["applicationId"] == "default-demo"
        assert pod.spec.scheduler_name == "yunikorn"


class TestKaiScheduler:
    def test_kai_routing_and_queue(self):
        from kuberay_amd.parallel import KaiBatchScheduler
        client = InMemoryClient()
        cluster = simple_raycluster("demo", gpus_per_worker=1)
        cluster.metadata.labels = {"kai.scheduler/queue": "team-a"}
        client.create(cluster)
        from kuberay_amd.common import pod as podlib
        group = cluster.spec.worker_group_specs[0]
        t = podlib.default_worker_pod_template(
            cluster, group, "p-", "x.default.svc.cluster.local", "6379")
        pod = podlib.build_pod(t, "worker", group.ray_start_params, "6379",
                               False, None, "x.default.svc.cluster.local")
        KaiBatchScheduler().add_metadata_to_pod(client, cluster,
                                                "default-group", pod)
        assert pod.spec.scheduler_name == "kai-scheduler"
        assert pod.metadata.labels["kai.scheduler/queue"] == "team-a"


class TestKubernetesWASScheduler:
    """Parity: batchscheduler/kubernetes-was/v1alpha2 (Workload + PodGroup
    gang scheduling the whole cluster, immutable-resource replacement)."""

    def _cluster(self, workers=3):
        cluster = simple_raycluster("was-demo", workers=workers,
                                    gpus_per_worker=1)
        cluster.metadata.labels = {"ray.io/gang-scheduling-enabled": "true"}
        return cluster

    def _sched(self):
        from kuberay_amd.parallel.batchscheduler import (
            KubernetesWASBatchScheduler)
        return KubernetesWASBatchScheduler()

    def test_creates_workload_and_podgroup(self):
        client = InMemoryClient()
        cluster = self._cluster(workers=3)
        cluster = client.create(cluster)
        sched = self._sched()
        sched.do_batch_scheduling_on_submission(client, cluster)
        wl = client.server.get("Workload", "default", "was-demo")
        tpl = wl["spec"]["podGroupTemplates"][0]
        assert tpl["name"] == "cluster"
        assert tpl["schedulingPolicy"]["gang"]["minCount"] == 4  # head+3
        pg = client.server.get("PodGroup", "default", "was-demo-cluster")
        assert pg["spec"]["podGroupTemplateRef"]["workload"] == {
            "workloadName": "was-demo", "podGroupTemplateName": "cluster"}
        assert pg["spec"]["schedulingPolicy"]["gang"]["minCount"] == 4
        # owned by the cluster for GC
        assert wl["metadata"]["ownerReferences"][0]["uid"] == \
            cluster.metadata.uid
        # idempotent second pass
        sched.do_batch_scheduling_on_submission(client, cluster)
        assert client.server.count("Workload") == 1

    def test_pod_metadata_sets_scheduling_group(self):
        client = InMemoryClient()
        cluster = self._cluster()
        cluster = client.create(cluster)
        sched = self._sched()
        group = cluster.spec.worker_group_specs[0]
        from kuberay_amd.common import pod as podlib
        fqdn = "x.default.svc.cluster.local"
        t = podlib.default_worker_pod_template(cluster, group, "p-", fqdn,
                                               "6379")
        pod = podlib.build_pod(t, "worker", group.ray_start_params, "6379",
                               False, None, fqdn)
        sched.add_metadata_to_pod(client, cluster, "default-group", pod)
        assert pod.spec.scheduler_name == "default-scheduler"
        assert pod.to_dict()["spec"]["schedulingGroup"] == {
            "podGroupName": "was-demo-cluster"}

    def test_min_count_change_replaces_immutable_resources(self):
        from kuberay_amd.parallel.batchscheduler import SchedulingRetry
        client = InMemoryClient()
        cluster = self._cluster(workers=3)
        cluster = client.create(cluster)
        sched = self._sched()
        sched.do_batch_scheduling_on_submission(client, cluster)
        # scale: minCount 4 -> 6; immutable resources must be recreated
        cluster.spec.worker_group_specs[0].replicas = 5
        client.update(cluster)
        with pytest.raises(SchedulingRetry):
            sched.do_batch_scheduling_on_submission(client, cluster)
        assert client.server.try_get("Workload", "default", "was-demo") is None
        # next reconcile recreates both at the new size
        sched.do_batch_scheduling_on_submission(client, cluster)
        pg = client.server.get("PodGroup", "default", "was-demo-cluster")
        assert pg["spec"]["schedulingPolicy"]["gang"]["minCount"] == 6

    def test_protection_finalizer_removed_before_delete(self):
        client = InMemoryClient()
        cluster = self._cluster()
        cluster = client.create(cluster)
        sched = self._sched()
        sched.do_batch_scheduling_on_submission(client, cluster)
        pg = client.server.get("PodGroup", "default", "was-demo-cluster")
        pg["metadata"]["finalizers"] = ["scheduling.k8s.io/podgroup-protection"]
        client.server.update(pg)
        sched.cleanup_on_completion(client, cluster)
        assert client.server.try_get("PodGroup", "default",
                                     "was-demo-cluster") is None
        assert client.server.try_get("Workload", "default", "was-demo") is None

    def test_skips_and_cleans_up_without_gang_label(self):
        client = InMemoryClient()
        cluster = self._cluster()
        cluster = client.create(cluster)
        sched = self._sched()
        sched.do_batch_scheduling_on_submission(client, cluster)
        assert client.server.count("Workload") == 1
        # label removed -> skip + stale resources cleaned up
        cluster.metadata.labels = {}
        client.update(cluster)
        sched.do_batch_scheduling_on_submission(client, cluster)
        assert client.server.count("Workload") == 0
        assert client.server.count("PodGroup") == 0

    def test_skips_when_autoscaling_enabled(self):
        client = InMemoryClient()
        cluster = self._cluster()
        cluster.spec.enable_in_tree_autoscaling = True
        cluster = client.create(cluster)
        sched = self._sched()
        sched.do_batch_scheduling_on_submission(client, cluster)
        assert client.server.count("Workload") == 0

    def test_foreign_workload_collision_fails_loudly(self):
        client = InMemoryClient()
        cluster = self._cluster()
        cluster = client.create(cluster)
        client.server.create({"apiVersion": "scheduling.k8s.io/v1alpha2",
                              "kind": "Workload",
                              "metadata": {"name": "was-demo",
                                           "namespace": "default"}})
        with pytest.raises(RuntimeError, match="not owned"):
            self._sched().do_batch_scheduling_on_submission(client, cluster)


class TestXgmiIslandScoring:
    """Best-fit island selection from node-labeller labels: gangs pin to the
    smallest island that fits; split-xGMI nodes only count their largest
    fully-connected island."""

    def _node(self, name, island, gpus, fully="true", largest=None):
        n = {"apiVersion": "v1", "kind": "Node",
             "metadata": {"name": name, "namespace": "default",
                          "labels": {
                              "amd.com/xgmi-island": island,
                              "amd.com/gpu.count": str(gpus),
                              "amd.com/xgmi-fully-connected": fully}}}
        if largest is not None:
            n["metadata"]["labels"]["amd.com/xgmi-largest-island"] = \
                str(largest)
        return n

    def test_best_fit_island_pinned(self):
        client = InMemoryClient()
        client.server.create(self._node("n1", "n1-island0", 8))
        client.server.create(self._node("n2", "n2-island0", 4))
        cluster = simple_raycluster("demo", workers=2, gpus_per_worker=2)
        cluster = client.create(cluster)
        sched = XgmiGangScheduler()
        pod = TestGangScheduling._pod(TestGangScheduling(), cluster)
        sched.add_metadata_to_pod(client, cluster, "default-group", pod)
        # demand = 2 pods x 2 GPUs = 4 -> best fit is the 4-GPU island
        terms = pod.spec.affinity["nodeAffinity"][
            "requiredDuringSchedulingIgnoredDuringExecution"][
            "nodeSelectorTerms"]
        assert terms[0]["matchExpressions"][0] == {
            "key": "amd.com/xgmi-island", "operator": "In",
            "values": ["n2-island0"]}

    def test_split_node_counts_largest_island_only(self):
        client = InMemoryClient()
        # split node: 8 GPUs but largest xGMI island is 4 -> cannot host a
        # 6-GPU gang; the healthy 8-GPU node must win
        client.server.create(self._node("split", "split-island0", 8,
                                        fully="false", largest=4))
        client.server.create(self._node("full", "full-island0", 8))
        cluster = simple_raycluster("demo", workers=3, gpus_per_worker=2)
        cluster = client.create(cluster)
        sched = XgmiGangScheduler()
        pod = TestGangScheduling._pod(TestGangScheduling(), cluster)
        sched.add_metadata_to_pod(client, cluster, "default-group", pod)
        terms = pod.spec.affinity["nodeAffinity"][
            "requiredDuringSchedulingIgnoredDuringExecution"][
            "nodeSelectorTerms"]
        assert terms[0]["matchExpressions"][0]["values"] == ["full-island0"]

    def test_gang_larger_than_every_island_refuses_not_splits(self):
        """Islands are known but none can hold the gang: every gang pod is
        pinned to the unschedulable sentinel so the gang holds Pending as a
        unit — splitting the RCCL ring across islands (PCIe) is never an
        outcome."""
        client = InMemoryClient()
        client.server.create(self._node("n1", "n1-island0", 2))
        cluster = simple_raycluster("demo", workers=4, gpus_per_worker=2)
        cluster = client.create(cluster)
        sched = XgmiGangScheduler()
        pod = TestGangScheduling._pod(TestGangScheduling(), cluster)
        sched.add_metadata_to_pod(client, cluster, "default-group", pod)
        terms = pod.spec.affinity["nodeAffinity"][
            "requiredDuringSchedulingIgnoredDuringExecution"][
            "nodeSelectorTerms"]
        assert terms[0]["matchExpressions"][0]["values"] == \
            [XgmiGangScheduler.UNSCHEDULABLE_ISLAND]
        assert pod.spec.affinity["podAffinity"]  # co-location still declared

    def test_no_nodes_no_pin(self):
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=2, gpus_per_worker=1)
        cluster = client.create(cluster)
        pod = TestGangScheduling._pod(TestGangScheduling(), cluster)
        XgmiGangScheduler().add_metadata_to_pod(client, cluster,
                                                "default-group", pod)
        # no island data at all (single-node dev): no hard pin, preferred
        # affinity alone keeps the gang together
        assert "nodeAffinity" not in pod.spec.affinity

    def test_mixed_cpu_gpu_groups_only_gpu_gets_gang(self):
        """A cluster with one CPU and one GPU worker group: only the GPU
        group's pods carry gang metadata; CPU pods schedule freely."""
        from kuberay_amd.models.raycluster import WorkerGroupSpec
        client = InMemoryClient()
        client.server.create(self._node("n1", "n1-island0", 8))
        cluster = simple_raycluster("demo", workers=2, gpus_per_worker=2)
        cluster.spec.worker_group_specs.append(WorkerGroupSpec.from_dict({
            "groupName": "cpu-group", "replicas": 2, "minReplicas": 0,
            "maxReplicas": 4, "rayStartParams": {},
            "template": {"spec": {"containers": [{
                "name": "ray-worker", "image": "img",
                "resources": {"limits": {"cpu": "4", "memory": "8Gi"}}}]}}}))
        cluster = client.create(cluster)
        sched = XgmiGangScheduler()
        gpu_pod = TestGangScheduling._pod(TestGangScheduling(), cluster)
        sched.add_metadata_to_pod(client, cluster, "default-group", gpu_pod)
        assert gpu_pod.metadata.labels.get("ray.io/xgmi-gang")

        from kuberay_amd.kube import objects as k8s
        cpu_pod = k8s.Pod(
            metadata=k8s.ObjectMeta(name="cpu-w", namespace="default"),
            spec=k8s.PodSpec(containers=[k8s.Container(
                name="ray-worker", image="img",
                resources=k8s.ResourceRequirements(
                    limits={"cpu": "4", "memory": "8Gi"}))]))
        sched.add_metadata_to_pod(client, cluster, "cpu-group", cpu_pod)
        assert not (cpu_pod.metadata.labels or {}).get("ray.io/xgmi-gang")
        assert cpu_pod.spec.affinity is None

    def test_pod_group_cleanup_on_completion(self):
        """cleanup_on_completion deletes the gang's PodGroup (volcano
        CleanupOnCompletion behavior), and is idempotent."""
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=2, gpus_per_worker=1)
        cluster = client.create(cluster)
        sched = XgmiGangScheduler()
        sched.do_batch_scheduling_on_submission(client, cluster)
        assert client.server.count("PodGroup") == 1
        sched.cleanup_on_completion(client, cluster)
        assert client.server.count("PodGroup") == 0
        sched.cleanup_on_completion(client, cluster)  # idempotent

    def test_rayjob_terminal_cleans_pod_group(self):
        """End-to-end: a RayJob whose cluster was gang-scheduled removes the
        PodGroup once the job is terminal."""
        import time as _time

        from kuberay_amd.models import RayJob
        from kuberay_amd.testing import ControlPlane
        cp = ControlPlane(kubelet_delay=0.01, job_runtime=0.1,
                          poll_seconds=0.05)
        sched = XgmiGangScheduler()
        cp.raycluster_reconciler.batch_scheduler = sched
        cp.rayjob_reconciler.batch_scheduler = sched
        cp.start()
        try:
            cluster_spec = simple_raycluster(
                "x", workers=1, gpus_per_worker=1).spec.to_dict()
            cp.client.create(RayJob.from_dict({
                "apiVersion": "ray.io/v1", "kind": "RayJob",
                "metadata": {"name": "gangjob", "namespace": "default"},
                "spec": {"entrypoint": "python x.py",
                         "rayClusterSpec": cluster_spec}}))
            assert cp.wait_for(lambda: cp.server.count("PodGroup") == 1,
                               timeout=15)
            assert cp.wait_for(
                lambda: (cp.client.get(RayJob, "default", "gangjob")
                         .status.job_deployment_status == "Complete"),
                timeout=25)
            assert cp.wait_for(lambda: cp.server.count("PodGroup") == 0,
                               timeout=15)
        finally:
            cp.stop()


class TestKubernetesWASEndToEnd:
    """e2ekuberneteswas analog: WAS scheduler wired into the real RayCluster
    reconciler; the immutable-replace retry path must converge under the
    workqueue's backoff."""

    def test_cluster_lifecycle_with_was_scheduling(self):
        import time as _time

        from kuberay_amd.models import RayCluster
        from kuberay_amd.parallel.batchscheduler import (
            KubernetesWASBatchScheduler)
        from kuberay_amd.testing import ControlPlane

        cp = ControlPlane(kubelet_delay=0.0, poll_seconds=0.05)
        cp.raycluster_reconciler.batch_scheduler = \
            KubernetesWASBatchScheduler()
        cp.start()
        try:
            rc = simple_raycluster("wase2e", workers=2, gpus_per_worker=1)
            rc.metadata.labels = {"ray.io/gang-scheduling-enabled": "true"}
            cp.client.create(rc)

            def wait(cond, timeout=20):
                deadline = _time.monotonic() + timeout
                while _time.monotonic() < deadline:
                    if cond():
                        return True
                    _time.sleep(0.05)
                return False

            assert wait(lambda: (cp.client.try_get(
                RayCluster, "default", "wase2e") or RayCluster()
                ).status.state == "ready")
            pg = cp.server.try_get("PodGroup", "default", "wase2e-cluster")
            assert pg["spec"]["schedulingPolicy"]["gang"]["minCount"] == 3
            # every pod joined the cluster PodGroup via spec.schedulingGroup
            pods = cp.server.list("Pod", "default")
            assert pods and all(
                p["spec"].get("schedulingGroup", {}).get("podGroupName")
                == "wase2e-cluster" for p in pods)

            # scale up: immutable resources get replaced (retry path), then
            # the cluster converges at the new size
            rc = cp.client.get(RayCluster, "default", "wase2e")
            rc.spec.worker_group_specs[0].replicas = 4
            cp.client.update(rc)
            assert wait(lambda: (pg := cp.server.try_get(
                "PodGroup", "default", "wase2e-cluster")) is not None
                and pg["spec"]["schedulingPolicy"]["gang"]["minCount"] == 5)
            assert wait(lambda: cp.client.get(
                RayCluster, "default", "wase2e"
                ).status.available_worker_replicas == 4)

            # deletion: owner GC removes Workload + PodGroup with the cluster
            cp.client.delete(cp.client.get(RayCluster, "default", "wase2e"))
            assert wait(lambda: cp.server.try_get(
                "Workload", "default", "wase2e") is None
                and cp.server.try_get("PodGroup", "default",
                                      "wase2e-cluster") is None)
        finally:
            cp.stop()


class TestIngressAndRoute:
    """common/ingress.go + common/openshift.go builder behavior."""

    def test_default_ingress_shape(self):
        from kuberay_amd.common.ingress import build_ingress_for_head_service
        cluster = simple_raycluster("demo")
        ing = build_ingress_for_head_service(cluster)
        rule = ing.spec["rules"][0]
        path = rule["http"]["paths"][0]
        assert path["path"] == "/demo/(.*)"
        assert path["pathType"] == "ImplementationSpecific"
        assert path["backend"]["service"]["port"]["number"] == 8265
        assert ing.metadata.annotations[
            "nginx.ingress.kubernetes.io/rewrite-target"] == "/$1"

    def test_ingress_options_override(self):
        from kuberay_amd.common.ingress import build_ingress_for_head_service
        cluster = simple_raycluster("demo")
        cluster.spec.head_group_spec.ingress_options = {
            "host": "ray.example.com", "path": "/dash",
            "ingressClassName": "nginx",
            "tls": [{"hosts": ["ray.example.com"]}]}
        ing = build_ingress_for_head_service(cluster)
        rule = ing.spec["rules"][0]
        assert rule["host"] == "ray.example.com"
        assert rule["http"]["paths"][0]["path"] == "/dash"
        assert rule["http"]["paths"][0]["pathType"] == "Exact"
        assert ing.spec["ingressClassName"] == "nginx"
        assert ing.spec["tls"]
        assert ing.metadata.annotations is None  # no rewrite with user path

    def test_openshift_route_shape(self):
        from kuberay_amd.common.openshift import build_route_for_head_service
        cluster = simple_raycluster("demo")
        route = build_route_for_head_service(cluster)
        assert route.kind == "Route"
        assert route.spec["to"]["name"].startswith("demo-head")
        assert route.spec["port"]["targetPort"] == 8265


def test_sharding_and_snapshot_flags_parse():
    from kuberay_amd.config import load_config
    cfg = load_config(["--shards", "4", "--shard-index", "2",
                       "--snapshot-interval", "5"])
    assert cfg.shards == 4
    assert cfg.shard_index == 2
    assert cfg.state_snapshot_interval_s == 5.0
    # defaults: unsharded
    assert load_config([]).shards == 1


def test_kubernetes_was_gate_selects_scheduler():
    """schedulermanager.go:56-60: the gate selects WAS; combining it with
    --batch-scheduler is a config error."""
    import kuberay_amd.features as features
    from kuberay_amd.config import Configuration
    from kuberay_amd.operator import build_manager
    from kuberay_amd.parallel.batchscheduler import KubernetesWASBatchScheduler
    features.set_gate("KubernetesWAS", True)
    try:
        m, _, _, _ = build_manager(Configuration(enable_metrics=False))
        rc = next(c for c in m.controllers if c.name == "raycluster")
        assert isinstance(rc.reconciler.batch_scheduler,
                          KubernetesWASBatchScheduler)
        with pytest.raises(ValueError, match="cannot be combined"):
            build_manager(Configuration(enable_metrics=False,
                                        enable_batch_scheduler=True,
                                        batch_scheduler="volcano"))
    finally:
        features.reset()


class TestMultiKueueInterop:
    """managedBy interop (reference raycluster_controller.go:158,
    rayjob_controller.go:107 + kueue multikueue semantics): a CR managed by
    an external controller is left entirely alone until the field points
    back at this operator."""

    MULTIKUEUE = "kueue.x-k8s.io/multikueue"

    def test_raycluster_managed_by_multikueue_not_reconciled(self,
                                                             control_plane):
        import time as _time
        rc = simple_raycluster("kueued", workers=2)
        rc.spec.managed_by = self.MULTIKUEUE
        control_plane.client.create(rc)
        _time.sleep(0.5)
        # no pods, no services, no status written
        assert control_plane.server.count("Pod") == 0
        assert control_plane.server.count("Service") == 0
        got = control_plane.client.get(RayCluster, "default", "kueued")
        assert got.status.state is None

    def test_rayjob_managed_by_multikueue_not_started(self, control_plane):
        import time as _time
        from kuberay_amd.models import RayJob
        from kuberay_amd.testing import simple_raycluster as src
        control_plane.client.create(RayJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayJob",
            "metadata": {"name": "kueued-job", "namespace": "default"},
            "spec": {"entrypoint": "python t.py",
                     "managedBy": self.MULTIKUEUE,
                     "rayClusterSpec": src("x").spec.to_dict()}}))
        _time.sleep(0.5)
        assert control_plane.server.count("RayCluster") == 0
        job = control_plane.client.get(RayJob, "default", "kueued-job")
        assert job.status.job_deployment_status in (None, "", "New")

    def test_handover_from_multikueue_starts_reconcile(self, control_plane):
        """Kueue admits the workload by flipping managedBy to the operator
        (multikueue handover): reconciliation must begin then."""
        rc = simple_raycluster("handover", workers=1)
        rc.spec.managed_by = self.MULTIKUEUE
        control_plane.client.create(rc)
        import time as _time
        _time.sleep(0.3)
        assert control_plane.server.count("Pod") == 0

        def admit(obj):
            obj.spec.managed_by = "ray.io/kuberay-operator"
        control_plane.client.update_with_retry(RayCluster, "default",
                                               "handover", admit)
        assert control_plane.wait_cluster_state("default", "handover",
                                                "ready", timeout=20)


class TestGateDefaultParity:
    """Pins every gate default to the reference table
    (pkg/features/features.go:104-117) so drift is caught at CI time."""

    REFERENCE_DEFAULTS = {
        "RayClusterStatusConditions": True,        # beta
        "RayJobDeletionPolicy": True,              # beta
        "RayMultiHostIndexing": True,              # beta
        "RayServiceIncrementalUpgrade": True,      # beta
        "RayCronJob": False,                       # alpha
        "SidecarSubmitterRestart": False,          # alpha
        "RayClusterNetworkPolicy": False,          # alpha
        "GCSFaultToleranceEmbeddedStorage": False, # alpha
        "RayClusterMTLS": False,                   # alpha
        "RayClusterHistoryServer": False,          # alpha
        "KubernetesWAS": False,                    # alpha
    }

    def test_defaults_match_reference(self):
        for gate, want in self.REFERENCE_DEFAULTS.items():
            assert features.enabled(gate) is want, gate


class TestNetworkPolicyModes:
    """networkpolicy_controller.go behavior matrix: deny modes, always-on
    intra-cluster + DNS allowances, user rule append, per-group overrides,
    stale policy GC on mode change."""

    def _cluster(self, mode, **np_extra):
        c = simple_raycluster("npx", workers=1)
        c.spec.network_policy = {"mode": mode, **np_extra}
        from kuberay_amd.models import RayCluster
        return RayCluster.from_dict(c.to_dict())

    def _reconcile(self, client, cluster):
        from kuberay_amd.ops.networkpolicy import NetworkPolicyReconciler
        NetworkPolicyReconciler(client).reconcile(
            ("default", cluster.metadata.name))

    def test_deny_all_ingress_only_sets_ingress_type(self):
        from kuberay_amd.ops.networkpolicy import build_head_network_policy
        p = build_head_network_policy(self._cluster("DenyAllIngress"))
        assert p.spec["policyTypes"] == ["Ingress"]
        assert "egress" not in p.spec
        # intra-cluster traffic always allowed
        sel = p.spec["ingress"][0]["from"][0]["podSelector"]["matchLabels"]
        assert sel["ray.io/cluster"] == "npx"

    def test_deny_all_egress_keeps_dns_open(self):
        from kuberay_amd.ops.networkpolicy import build_head_network_policy
        p = build_head_network_policy(self._cluster("DenyAllEgress"))
        assert p.spec["policyTypes"] == ["Egress"]
        ports = [pt["port"] for r in p.spec["egress"]
                 for pt in r.get("ports", [])]
        assert ports.count(53) == 2  # UDP + TCP DNS

    def test_user_rules_appended_after_builtins(self):
        from kuberay_amd.ops.networkpolicy import build_head_network_policy
        p = build_head_network_policy(self._cluster(
            "DenyAll", head={"ingressRules": [
                {"from": [{"ipBlock": {"cidr": "10.0.0.0/8"}}]}]}))
        assert p.spec["ingress"][-1]["from"][0]["ipBlock"][
            "cidr"] == "10.0.0.0/8"

    def test_per_group_override_layers_on_general_policy(self):
        from kuberay_amd.ops.networkpolicy import (
            build_worker_network_policies)
        ps = build_worker_network_policies(self._cluster(
            "DenyAll", workerGroups=[{"groupName": "default-group",
                                      "egressRules": [
                {"to": [{"ipBlock": {"cidr": "192.168.0.0/16"}}]}]}]))
        names = [p.metadata.name for p in ps]
        assert names == ["npx-workers", "npx-workers-default-group"]
        grp = ps[1].spec["podSelector"]["matchLabels"]
        assert grp["ray.io/group"] == "default-group"

    def test_stale_policy_gc_on_mode_change(self):
        from kuberay_amd.kube import objects as k8s
        from kuberay_amd.kube.client import InMemoryClient
        client = InMemoryClient()
        c = self._cluster("DenyAll", workerGroups=[
            {"groupName": "default-group"}])
        client.create(c)
        self._reconcile(client, c)
        assert len(client.list(k8s.NetworkPolicy, "default")) == 3
        # drop the per-group override: its policy must be GC'd
        c2 = self._cluster("DenyAll")
        c2.metadata.resource_version = None
        from kuberay_amd.models import RayCluster
        client.update_with_retry(
            RayCluster, "default", "npx",
            lambda cur: setattr(cur.spec, "network_policy",
                                c2.spec.network_policy))
        self._reconcile(client, client.get(RayCluster, "default", "npx"))
        left = {p.metadata.name
                for p in client.list(k8s.NetworkPolicy, "default")}
        assert left == {"npx-head", "npx-workers"}
