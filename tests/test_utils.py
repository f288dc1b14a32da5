"""Utils unit tests: names, quantities, resources, hashing, validation
(reference analogs: utils/util_test.go, utils/validation_test.go)."""
import pytest

from kuberay_amd.models import RayCluster, RayJob
from kuberay_amd.testing import simple_raycluster
from kuberay_amd.utils import constants as C
from kuberay_amd.utils import names
from kuberay_amd.utils.hashing import hash_without_replicas_and_workers_to_delete
from kuberay_amd.utils.quantity import add_quantities, format_quantity, parse_quantity
from kuberay_amd.utils.resources import (
    calculate_desired_replicas,
    calculate_desired_resources,
    calculate_max_replicas,
    calculate_min_replicas,
    container_gpu_count,
    is_amd_gpu_resource,
    worker_group_desired_replicas,
)
from kuberay_amd.utils.validation import (
    validate_raycluster_spec,
    validate_rayjob_spec,
    validate_rayservice_spec,
)


class TestNames:
    def test_check_name_truncates_from_front(self):
        long = "a" * 60
        out = names.check_name(long)
        assert len(out) == 50

    def test_check_name_fixes_leading_digit(self):
        assert names.check_name("9abc") == "rabc"

    def test_pod_name(self):
        assert names.pod_name("demo", "head", True) == "demo-head-"
        assert names.pod_name("demo", "worker", False) == "demo-worker"
        long = "x" * 60
        assert len(names.pod_name(long, "worker", True)) == 50 + len("-worker-")

    def test_head_service_name(self):
        cluster = simple_raycluster("demo")
        assert names.head_service_name("RayCluster", cluster.spec, "demo") == "demo-head-svc"
        assert names.head_service_name("RayService", cluster.spec, "svc") == "svc-head-svc"

    def test_fqdn(self):
        cluster = simple_raycluster("demo")
        assert names.fqdn_service_name(cluster, "ns1") == \
            "demo-head-svc.ns1.svc.cluster.local"
        assert names.extract_ray_ip_from_fqdn("a.b.svc.cluster.local") == "a"

    def test_check_label(self):
        assert len(names.check_label("a" * 80)) == 63


class TestQuantity:
    @pytest.mark.parametrize("value,expected", [
        ("500m", 0.5), ("2", 2), ("2Gi", 2 * 1024**3), ("1k", 1000),
        ("100Mi", 100 * 1024**2), (None, 0), ("0.5", 0.5), ("1e3", 1000),
    ])
    def test_parse(self, value, expected):
        assert float(parse_quantity(value)) == expected

    def test_format_roundtrip(self):
        assert format_quantity(parse_quantity("2Gi")) == "2Gi"
        assert format_quantity(parse_quantity("500m")) == "500m"
        assert format_quantity(parse_quantity("3")) == "3"

    def test_add(self):
        assert add_quantities("500m", "1500m") == "2"
        assert add_quantities("1Gi", "1Gi") == "2Gi"


class TestResources:
    def test_amd_gpu_only(self):
        assert is_amd_gpu_resource("amd.com/gpu")
        assert not is_amd_gpu_resource("nvidia.com/gpu")
        assert not is_amd_gpu_resource("nvidia.com/mig-1g.5gb")
        assert not is_amd_gpu_resource("google.com/tpu")

    def test_worker_group_replicas_clamped(self):
        cluster = simple_raycluster("demo", workers=5)
        g = cluster.spec.worker_group_specs[0]
        g.min_replicas, g.max_replicas = 2, 3
        assert worker_group_desired_replicas(g) == 3
        g.replicas = 1
        assert worker_group_desired_replicas(g) == 2
        g.suspend = True
        assert worker_group_desired_replicas(g) == 0

    def test_desired_counts_num_of_hosts(self):
        cluster = simple_raycluster("demo", workers=2, num_of_hosts=4)
        assert calculate_desired_replicas(cluster) == 8

    def test_min_max(self):
        cluster = simple_raycluster("demo", workers=2)
        g = cluster.spec.worker_group_specs[0]
        g.min_replicas, g.max_replicas = 1, 10
        assert calculate_min_replicas(cluster) == 1
        assert calculate_max_replicas(cluster) == 10

    def test_desired_resources_sums_head_and_workers(self):
        cluster = simple_raycluster("demo", workers=2, gpus_per_worker=2)
        totals = calculate_desired_resources(cluster)
        assert totals["desiredGPU"] == "4"
        assert totals["desiredCPU"] == "3"  # head 1 + 2 workers * 1


class TestHashing:
    def test_stable_under_scale_fields(self):
        cluster = simple_raycluster("demo")
        h1 = hash_without_replicas_and_workers_to_delete(cluster.spec)
        cluster.spec.worker_group_specs[0].replicas = 99
        cluster.spec.worker_group_specs[0].min_replicas = 9
        cluster.spec.worker_group_specs[0].scale_strategy.workers_to_delete = ["a"]
        assert hash_without_replicas_and_workers_to_delete(cluster.spec) == h1

    def test_changes_on_template_change(self):
        cluster = simple_raycluster("demo")
        h1 = hash_without_replicas_and_workers_to_delete(cluster.spec)
        cluster.spec.worker_group_specs[0].template.spec.containers[0].image = "z"
        assert hash_without_replicas_and_workers_to_delete(cluster.spec) != h1

    def test_tolerations_muted(self):
        cluster = simple_raycluster("demo")
        h1 = hash_without_replicas_and_workers_to_delete(cluster.spec)
        cluster.spec.worker_group_specs[0].template.spec.tolerations = [
            {"key": "k", "operator": "Exists"}]
        assert hash_without_replicas_and_workers_to_delete(cluster.spec) == h1


class TestValidation:
    def test_valid_cluster(self):
        assert validate_raycluster_spec(simple_raycluster("demo")) == []

    def test_min_gt_max(self):
        c = simple_raycluster("demo")
        c.spec.worker_group_specs[0].min_replicas = 5
        c.spec.worker_group_specs[0].max_replicas = 2
        assert any("minReplicas" in e for e in validate_raycluster_spec(c))

    def test_duplicate_group_names(self):
        c = simple_raycluster("demo")
        c.spec.worker_group_specs.append(c.spec.worker_group_specs[0].clone())
        assert any("duplicate" in e for e in validate_raycluster_spec(c))

    def test_gcs_ft_embedded_rejects_redis_fields(self):
        c = simple_raycluster("demo", gcsFaultToleranceOptions={
            "backend": "embedded", "redisAddress": "redis://x"})
        assert any("embedded" in e for e in validate_raycluster_spec(c))

    def test_idle_timeout_requires_v2(self):
        c = simple_raycluster("demo")
        c.spec.worker_group_specs[0].idle_timeout_seconds = 60
        assert any("v2" in e for e in validate_raycluster_spec(c))
        c.spec.autoscaler_options = type(c.spec).model_fields["autoscaler_options"] \
            .annotation.__args__[0](version="v2")
        assert validate_raycluster_spec(c) == []

    def test_rayjob_requires_cluster_source(self):
        job = RayJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayJob",
            "metadata": {"name": "j"},
            "spec": {"entrypoint": "x"}})
        assert any("rayClusterSpec" in e for e in validate_rayjob_spec(job))

    def test_rayjob_ttl_requires_shutdown(self):
        job = RayJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayJob",
            "metadata": {"name": "j"},
            "spec": {"entrypoint": "x", "ttlSecondsAfterFinished": 10,
                     "rayClusterSpec": simple_raycluster("x").spec.to_dict()}})
        assert any("shutdownAfterJobFinishes" in e for e in validate_rayjob_spec(job))

    def test_rayjob_deletion_rules_exclusive(self):
        job = RayJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayJob",
            "metadata": {"name": "j"},
            "spec": {"entrypoint": "x",
                     "rayClusterSpec": simple_raycluster("x").spec.to_dict(),
                     "deletionStrategy": {
                         "onSuccess": {"policy": "DeleteCluster"},
                         "onFailure": {"policy": "DeleteNone"},
                         "deletionRules": [{"policy": "DeleteSelf",
                                            "condition": {"jobStatus": "FAILED"}}]}}})
        assert any("simultaneously" in e for e in validate_rayjob_spec(job))

    def test_rayservice_incremental_requires_gateway(self):
        from kuberay_amd.models import RayService
        svc = RayService.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayService",
            "metadata": {"name": "s"},
            "spec": {
                "serveConfigV2": "applications:\n- name: a\n",
                "rayClusterConfig": simple_raycluster("x").spec.to_dict(),
                "upgradeStrategy": {"type": "NewClusterWithIncrementalUpgrade"}}})
        errs = validate_rayservice_spec(svc)
        assert any("ClusterUpgradeOptions are required" in e for e in errs)


class TestStructuralSchemas:
    def test_all_crds_structural_and_clean(self):
        import json
        from kuberay_amd.crds import all_crds
        for crd in all_crds():
            schema = crd["spec"]["versions"][0]["schema"]["openAPIV3Schema"]
            text = json.dumps(schema)
            for banned in ("$ref", "$defs", "anyOf", "allOf", "oneOf"):
                assert banned not in text, (crd["metadata"]["name"], banned)
            assert schema["type"] == "object"
            assert "spec" in schema["properties"]
            assert "status" in schema["properties"]

    def test_samples_validate_against_schema(self):
        """Every shipped sample must pass its CRD's structural schema
        (jsonschema-lite check on typed fields)."""
        import glob
        import os
        import yaml as _yaml
        from kuberay_amd.crds import all_crds
        schemas = {c["spec"]["names"]["kind"]:
                   c["spec"]["versions"][0]["schema"]["openAPIV3Schema"]
                   for c in all_crds()}

        def check(node, schema, path):
            if schema.get("x-kubernetes-preserve-unknown-fields"):
                return
            t = schema.get("type")
            if t == "object" and isinstance(node, dict):
                props = schema.get("properties", {})
                for k, v in node.items():
                    if k in props:
                        check(v, props[k], f"{path}.{k}")
            elif t == "array" and isinstance(node, list):
                for i, v in enumerate(node):
                    check(v, schema.get("items", {}), f"{path}[{i}]")
            elif t == "integer":
                assert isinstance(node, int), (path, node)
            elif t == "string":
                assert isinstance(node, str), (path, node)
            elif t == "boolean":
                assert isinstance(node, bool), (path, node)

        samples = os.path.join(os.path.dirname(__file__), "..", "deploy", "samples")
        for f in sorted(glob.glob(os.path.join(samples, "*.yaml"))):
            for doc in _yaml.safe_load_all(open(f)):
                if doc and doc.get("kind") in schemas:
                    check(doc, schemas[doc["kind"]], os.path.basename(f))


class TestDeletionStrategyValidation:
    """validation.go:543-967 deletion-rule branches."""

    def _job(self, ds, **spec):
        from kuberay_amd.models import RayJob
        from kuberay_amd.testing import simple_raycluster
        body = {"entrypoint": "python x.py",
                "rayClusterSpec": simple_raycluster("x").spec.to_dict(),
                "deletionStrategy": ds}
        body.update(spec)
        return RayJob.from_dict({
            "apiVersion": "ray.io/v1", "kind": "RayJob",
            "metadata": {"name": "j", "namespace": "default"},
            "spec": body})

    def test_legacy_and_rules_mutually_exclusive(self):
        from kuberay_amd.utils.validation import validate_rayjob_spec
        job = self._job({"onSuccess": {"policy": "DeleteCluster"},
                         "onFailure": {"policy": "DeleteNone"},
                         "deletionRules": [{"policy": "DeleteCluster",
                                            "condition": {"jobStatus":
                                                          "SUCCEEDED"}}]})
        assert any("simultaneously" in e for e in
                   validate_rayjob_spec(job))

    def test_legacy_requires_both_blocks(self):
        from kuberay_amd.utils.validation import validate_rayjob_spec
        job = self._job({"onSuccess": {"policy": "DeleteCluster"}})
        assert any("OnSuccess and DeletionStrategy.OnFailure" in e for e in
                   validate_rayjob_spec(job))

    def test_non_terminal_job_status_rejected(self):
        from kuberay_amd.utils.validation import validate_rayjob_spec
        job = self._job({"deletionRules": [{
            "policy": "DeleteCluster",
            "condition": {"jobStatus": "RUNNING"}}]})
        assert any("not terminal" in e for e in validate_rayjob_spec(job))

    def test_bad_policy_rejected(self):
        from kuberay_amd.utils.validation import validate_rayjob_spec
        job = self._job({"deletionRules": [{
            "policy": "ExplodeCluster",
            "condition": {"jobStatus": "SUCCEEDED"}}]})
        assert any("invalid policy" in e for e in validate_rayjob_spec(job))

    def test_rules_conflict_with_shutdown_flag(self):
        from kuberay_amd.utils.validation import validate_rayjob_spec
        job = self._job({"deletionRules": [{
            "policy": "DeleteCluster",
            "condition": {"jobStatus": "SUCCEEDED"}}]},
            shutdownAfterJobFinishes=True)
        assert any("shutdownAfterJobFinishes" in e for e in
                   validate_rayjob_spec(job))

    def test_valid_rules_pass(self):
        from kuberay_amd.utils.validation import validate_rayjob_spec
        job = self._job({"deletionRules": [
            {"policy": "DeleteCluster",
             "condition": {"jobStatus": "SUCCEEDED", "ttlSeconds": 60}},
            {"policy": "DeleteNone",
             "condition": {"jobDeploymentStatus": "Failed"}}]},
            shutdownAfterJobFinishes=False)
        assert validate_rayjob_spec(job) == []


class TestTLSValidation:
    """validation.go:446-480: user-set TLS env/mounts conflict with
    operator-managed TLS."""

    @pytest.fixture(autouse=True)
    def _gate(self):
        import kuberay_amd.features as features
        features.set_gate("RayClusterMTLS", True)
        yield
        features.reset()

    def _cluster(self, env=None, mount=None):
        from kuberay_amd.testing import simple_raycluster
        rc = simple_raycluster("tlsv")
        rc.spec.tls_options = {"enabled": True}
        head = rc.spec.head_group_spec.template.spec.containers[0]
        if env:
            from kuberay_amd.kube.objects import EnvVar
            head.env = (head.env or []) + [EnvVar(name=env, value="1")]
        if mount:
            from kuberay_amd.kube.objects import VolumeMount
            head.volume_mounts = (head.volume_mounts or []) + [
                VolumeMount(name="user-tls", mount_path=mount)]
        return rc

    def test_user_tls_env_rejected(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        errs = validate_raycluster_spec(self._cluster(env="RAY_USE_TLS"))
        assert any("RAY_USE_TLS" in e for e in errs)

    def test_user_cert_mount_rejected(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        errs = validate_raycluster_spec(
            self._cluster(mount="/etc/ray/tls"))
        assert any("cert mount" in e for e in errs)

    def test_clean_tls_cluster_passes(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        assert validate_raycluster_spec(self._cluster()) == []

    def test_env_fine_when_tls_disabled(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        rc = self._cluster(env="RAY_USE_TLS")
        rc.spec.tls_options = None
        assert validate_raycluster_spec(rc) == []


class TestGroupResourcesLabelsValidation:
    """validation.go:60-101: top-level resources/labels vs rayStartParams."""

    def _cluster(self, params=None, resources=None, labels=None):
        from kuberay_amd.testing import simple_raycluster
        rc = simple_raycluster("grl")
        g = rc.spec.worker_group_specs[0]
        if params:
            g.ray_start_params.update(params)
        if resources is not None:
            g.resources = resources
        if labels is not None:
            g.labels = labels
        return rc

    def test_conflicting_resources_rejected(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        errs = validate_raycluster_spec(self._cluster(
            params={"num-gpus": "4"}, resources={"CPU": "8"}))
        assert any("use only one" in e for e in errs)

    def test_ray_start_labels_param_rejected(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        errs = validate_raycluster_spec(self._cluster(
            params={"labels": "a=b"}))
        assert any("top-level labels field" in e for e in errs)

    def test_bad_label_syntax_rejected(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        errs = validate_raycluster_spec(self._cluster(
            labels={"-bad-key": "x"}))
        assert any("invalid label key" in e for e in errs)
        errs = validate_raycluster_spec(self._cluster(
            labels={"ok": "bad value!"}))
        assert any("invalid label value" in e for e in errs)

    def test_clean_group_labels_pass(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        assert validate_raycluster_spec(self._cluster(
            resources={"TPUAnalog": "0"},
            labels={"ray.io/market-type": "spot"})) == []


class TestClusterUpgradeAndPriorityValidation:
    """validation.go:42-58 + :1113-1142 + :651-678."""

    def test_invalid_upgrade_type_rejected(self):
        from kuberay_amd.testing import simple_raycluster
        from kuberay_amd.utils.validation import validate_raycluster_spec
        rc = simple_raycluster("u1")
        from kuberay_amd.models.raycluster import RayClusterUpgradeStrategy
        rc.spec.upgrade_strategy = RayClusterUpgradeStrategy(type="BlueGreen")
        assert any("upgradeStrategy.type" in e
                   for e in validate_raycluster_spec(rc))

    def test_upgrade_on_child_cluster_rejected(self):
        from kuberay_amd.testing import simple_raycluster
        from kuberay_amd.utils.validation import validate_raycluster_spec
        rc = simple_raycluster("u2")
        from kuberay_amd.models.raycluster import RayClusterUpgradeStrategy
        rc.spec.upgrade_strategy = RayClusterUpgradeStrategy(type="Recreate")
        rc.metadata.labels = {"ray.io/originated-from-crd": "RayJob"}
        assert any("created by RayJob" in e
                   for e in validate_raycluster_spec(rc))

    def test_priority_needs_new_ray_and_autoscaler_v2(self):
        from kuberay_amd.testing import simple_raycluster
        from kuberay_amd.utils.validation import validate_raycluster_spec
        rc = simple_raycluster("p1")
        rc.spec.worker_group_specs[0].priority = 5
        rc.spec.ray_version = "2.46.0"
        errs = validate_raycluster_spec(rc)
        assert any("requires Ray >= 2.56.0" in e for e in errs)
        rc.spec.ray_version = "2.56.0"
        errs = validate_raycluster_spec(rc)
        assert any("autoscaler v2" in e for e in errs)
        from kuberay_amd.models.raycluster import AutoscalerOptions
        rc.spec.autoscaler_options = AutoscalerOptions(version="v2")
        assert validate_raycluster_spec(rc) == []

    def test_initializing_timeout_annotation(self):
        from kuberay_amd.models import RayService
        from kuberay_amd.testing import simple_raycluster
        from kuberay_amd.utils.validation import validate_rayservice_spec

        def svc(value):
            return RayService.from_dict({
                "apiVersion": "ray.io/v1", "kind": "RayService",
                "metadata": {"name": "s", "namespace": "default",
                             "annotations": {
                                 "ray.io/initializing-timeout": value}},
                "spec": {"serveConfigV2": "applications: []",
                         "rayClusterConfig":
                             simple_raycluster("x").spec.to_dict()}})

        assert validate_rayservice_spec(svc("30m")) == []
        assert validate_rayservice_spec(svc("600")) == []
        assert any("invalid format" in e
                   for e in validate_rayservice_spec(svc("soon")))
        assert any("must be positive" in e
                   for e in validate_rayservice_spec(svc("0")))


class TestCollectorOptionsValidation:
    """validation.go:1029-1064: collector env hygiene."""

    @pytest.fixture(autouse=True)
    def _gate(self):
        import kuberay_amd.features as features
        features.set_gate("RayClusterHistoryServer", True)
        yield
        features.reset()

    def _cluster(self, env):
        from kuberay_amd.testing import simple_raycluster
        return simple_raycluster("col", historyServerOptions={
            "collectorOptions": {"image": "rocm/history-collector:1.0",
                                 "env": env}})

    def test_managed_env_rejected(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        errs = validate_raycluster_spec(self._cluster(
            [{"name": "POD_IP", "value": "1.2.3.4"}]))
        assert any("must not contain POD_IP" in e for e in errs)

    def test_cloud_backend_requires_companion_env(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        errs = validate_raycluster_spec(self._cluster(
            [{"name": "STORAGE_BACKEND", "value": "s3"}]))
        assert any("S3_REGION" in e for e in errs)
        errs = validate_raycluster_spec(self._cluster(
            [{"name": "STORAGE_BACKEND", "value": "s3"},
             {"name": "S3_REGION", "value": "us-east-1"}]))
        assert errs == []

    def test_local_backend_passes(self):
        from kuberay_amd.utils.validation import validate_raycluster_spec
        assert validate_raycluster_spec(self._cluster(
            [{"name": "STORAGE_BACKEND", "value": "local"}])) == []
