"""APIServer v1/v2 + HttpKubeClient + kray CLI tests
(reference analogs: apiserver unit/e2e tests, kubectl-plugin e2e suites)."""
import pytest
from click.testing import CliRunner
from fastapi.testclient import TestClient

from kuberay_amd.apiserver import create_app
from kuberay_amd.cli.main import cli
from kuberay_amd.kube.client import InMemoryClient
from kuberay_amd.kube.httpclient import HttpKubeClient
from kuberay_amd.models import RayCluster, RayJob
from kuberay_amd.utils.fake_dashboard import FakeRayDashboardClient

CLUSTER_BODY = {
    "name": "c1", "version": "2.46.0", "user": "alice@example.com",
    "clusterSpec": {
        "headGroupSpec": {"computeTemplate": "tpl", "rayStartParams": {}},
        "workerGroupSpec": [{
            "groupName": "g", "computeTemplate": "tpl",
            "replicas": 2, "minReplicas": 0, "maxReplicas": 4}],
    },
}


@pytest.fixture()
def api():
    client = InMemoryClient()
    fake = FakeRayDashboardClient()
    app = create_app(client, dashboard_factory=lambda name: fake)
    t = TestClient(app)
    t.post("/apis/v1/namespaces/ns1/compute_templates",
           json={"name": "tpl", "cpu": 4, "memory": 8, "gpu": 2})
    return t, client, fake


class TestComputeTemplates:
    def test_crud(self, api):
        t, _, _ = api
        r = t.get("/apis/v1/namespaces/ns1/compute_templates")
        assert r.status_code == 200
        tpl = r.json()["computeTemplates"][0]
        assert tpl["gpuAccelerator"] == "amd.com/gpu"
        assert t.get("/apis/v1/namespaces/ns1/compute_templates/tpl").status_code == 200
        assert t.delete("/apis/v1/namespaces/ns1/compute_templates/tpl").status_code == 200
        assert t.get("/apis/v1/namespaces/ns1/compute_templates/tpl").status_code == 404


class TestClusterService:
    def test_create_expands_template_to_amd_gpu(self, api):
        t, client, _ = api
        r = t.post("/apis/v1/namespaces/ns1/clusters", json=CLUSTER_BODY)
        assert r.status_code == 200, r.text
        rc = client.get(RayCluster, "ns1", "c1")
        limits = rc.spec.worker_group_specs[0].template.spec.containers[0] \
            .resources.limits
        assert limits["amd.com/gpu"] == "2"
        assert limits["cpu"] == "4"
        # user label sanitized
        assert rc.metadata.labels["ray.io/user"] == "alice-example.com"

    def test_get_list_delete(self, api):
        t, _, _ = api
        t.post("/apis/v1/namespaces/ns1/clusters", json=CLUSTER_BODY)
        assert len(t.get("/apis/v1/namespaces/ns1/clusters").json()["clusters"]) == 1
        assert t.get("/apis/v1/namespaces/ns1/clusters/c1").status_code == 200
        assert t.delete("/apis/v1/namespaces/ns1/clusters/c1").status_code == 200
        assert t.get("/apis/v1/namespaces/ns1/clusters/c1").status_code == 404

    def test_invalid_rejected(self, api):
        t, _, _ = api
        bad = dict(CLUSTER_BODY)
        bad["clusterSpec"] = {
            "headGroupSpec": {"computeTemplate": "tpl"},
            "workerGroupSpec": [{"groupName": "g", "computeTemplate": "tpl",
                                 "replicas": 2, "minReplicas": 5, "maxReplicas": 1}]}
        r = t.post("/apis/v1/namespaces/ns1/clusters", json=bad)
        assert r.status_code == 400


class TestJobService:
    def test_create_and_get(self, api):
        t, client, _ = api
        r = t.post("/apis/v1/namespaces/ns1/jobs", json={
            "name": "j1", "entrypoint": "python x.py",
            "clusterSpec": CLUSTER_BODY["clusterSpec"]})
        assert r.status_code == 200, r.text
        job = client.get(RayJob, "ns1", "j1")
        assert job.spec.entrypoint == "python x.py"
        assert job.spec.ray_cluster_spec is not None
        assert t.get("/apis/v1/namespaces/ns1/jobs/j1").json()["name"] == "j1"


class TestJobSubmissionProxy:
    def test_submit_and_get(self, api):
        t, _, fake = api
        t.post("/apis/v1/namespaces/ns1/clusters", json=CLUSTER_BODY)
        r = t.post("/apis/v1/namespaces/ns1/jobsubmissions/c1",
                   json={"entrypoint": "python x.py", "submission_id": "sub-1"})
        assert r.status_code == 200
        assert r.json()["submissionId"] == "sub-1"
        r = t.get("/apis/v1/namespaces/ns1/jobsubmissions/c1/sub-1")
        assert r.status_code == 200


class TestV2Proxy:
    def test_restricted_to_ray_resources(self, api):
        t, _, _ = api
        assert t.get("/apis/ray.io/v1/namespaces/ns1/pods").status_code == 404

    def test_full_cr_roundtrip(self, api):
        t, _, _ = api
        from kuberay_amd.testing import simple_raycluster
        body = simple_raycluster("v2c").to_dict()
        r = t.post("/apis/ray.io/v1/namespaces/ns1/rayclusters", json=body)
        assert r.status_code == 200
        got = t.get("/apis/ray.io/v1/namespaces/ns1/rayclusters/v2c").json()
        assert got["spec"]["workerGroupSpecs"][0]["groupName"] == "default-group"


@pytest.fixture()
def http_stack(api):
    """HttpKubeClient wired to the FastAPI app via in-process transport."""
    t, client, fake = api
    hc = HttpKubeClient("http://testserver", http_client=t)
    return hc, client


class TestHttpKubeClient:
    def test_crud(self, http_stack):
        hc, backing = http_stack
        from kuberay_amd.testing import simple_raycluster
        created = hc.create(simple_raycluster("hc1", namespace="ns1"))
        assert created.metadata.uid
        got = hc.get(RayCluster, "ns1", "hc1")
        got.spec.worker_group_specs[0].replicas = 5
        hc.update(got)
        assert backing.get(RayCluster, "ns1", "hc1") \
            .spec.worker_group_specs[0].replicas == 5
        assert len(hc.list(RayCluster, "ns1")) == 1
        hc.delete(RayCluster, "ns1", "hc1")
        assert hc.try_get(RayCluster, "ns1", "hc1") is None


@pytest.fixture()
def kray(api, monkeypatch):
    t, client, _ = api
    hc = HttpKubeClient("http://testserver", http_client=t)
    import importlib
    climod = importlib.import_module("kuberay_amd.cli.main")
    monkeypatch.setattr(climod, "make_client", lambda server: hc)
    return CliRunner(), client


class TestKrayCli:
    def test_version(self, kray):
        runner, _ = kray
        r = runner.invoke(cli, ["version"])
        assert r.exit_code == 0 and "kuberay-amd" in r.output

    def test_create_get_scale_delete_cluster(self, kray):
        runner, backing = kray
        r = runner.invoke(cli, ["-n", "ns1", "create", "cluster", "k1",
                                "--worker-replicas", "2", "--worker-gpu", "1"])
        assert r.exit_code == 0, r.output
        rc = backing.get(RayCluster, "ns1", "k1")
        limits = rc.spec.worker_group_specs[0].template.spec.containers[0] \
            .resources.limits
        assert limits["amd.com/gpu"] == "1"

        r = runner.invoke(cli, ["-n", "ns1", "get", "cluster"])
        assert r.exit_code == 0 and "k1" in r.output

        r = runner.invoke(cli, ["-n", "ns1", "scale", "cluster", "k1",
                                "--replicas", "4"])
        assert r.exit_code == 0, r.output
        assert backing.get(RayCluster, "ns1", "k1") \
            .spec.worker_group_specs[0].replicas == 4

        r = runner.invoke(cli, ["-n", "ns1", "delete", "cluster", "k1"])
        assert r.exit_code == 0
        assert backing.try_get(RayCluster, "ns1", "k1") is None

    def test_create_workergroup(self, kray):
        runner, backing = kray
        runner.invoke(cli, ["-n", "ns1", "create", "cluster", "k1"])
        r = runner.invoke(cli, ["-n", "ns1", "create", "workergroup", "k1",
                                "--group-name", "gpu-group", "--worker-gpu", "8"])
        assert r.exit_code == 0, r.output
        rc = backing.get(RayCluster, "ns1", "k1")
        assert [g.group_name for g in rc.spec.worker_group_specs] == \
            ["default-group", "gpu-group"]

    def test_job_submit(self, kray):
        runner, backing = kray
        r = runner.invoke(cli, ["-n", "ns1", "job", "submit", "--name", "j1",
                                "--entrypoint", "python t.py",
                                "--worker-gpu", "2"])
        assert r.exit_code == 0, r.output
        job = backing.get(RayJob, "ns1", "j1")
        assert job.spec.entrypoint == "python t.py"

    def test_dry_run_yaml(self, kray):
        runner, backing = kray
        r = runner.invoke(cli, ["-n", "ns1", "create", "cluster", "k1",
                                "--worker-gpu", "4", "--dry-run"])
        assert r.exit_code == 0
        assert "amd.com/gpu: '4'" in r.output
        assert backing.try_get(RayCluster, "ns1", "k1") is None


class TestGrpcApi:
    @pytest.fixture()
    def grpc_stack(self):
        import grpc
        from kuberay_amd.apiserver.grpc_api import (
            Cluster, ComputeTemplate, DeleteRequest, GetRequest, ListRequest,
            ListClusterResponse, create_grpc_server)
        from kuberay_amd.kube.client import InMemoryClient
        client = InMemoryClient()
        server = create_grpc_server(client, port=0)
        port = server.add_insecure_port("127.0.0.1:0")
        server.start()
        channel = grpc.insecure_channel(f"127.0.0.1:{port}")
        yield channel, client
        server.stop(0)

    def _call(self, channel, service, method, request, resp_cls):
        import grpc
        fn = channel.unary_unary(
            f"/kuberayamd.v1.{service}/{method}",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=resp_cls.FromString)
        return fn(request, timeout=5)

    def test_cluster_crud_over_grpc(self, grpc_stack):
        import json
        import grpc
        from kuberay_amd.apiserver.grpc_api import (
            Cluster, ComputeTemplate, DeleteRequest, Empty, GetRequest,
            ListRequest, ListClusterResponse)
        channel, backing = grpc_stack
        self._call(channel, "ComputeTemplateService", "CreateComputeTemplate",
                   ComputeTemplate(name="tpl", namespace="ns1", cpu=4,
                                   memory=8, gpu=2), ComputeTemplate)
        spec = {"headGroupSpec": {"computeTemplate": "tpl"},
                "workerGroupSpec": [{"groupName": "g", "computeTemplate": "tpl",
                                     "replicas": 1, "maxReplicas": 2}]}
        created = self._call(channel, "ClusterService", "CreateCluster",
                             Cluster(name="gc1", namespace="ns1",
                                     version="2.46.0",
                                     spec_json=json.dumps(spec)), Cluster)
        assert created.name == "gc1"
        # expanded to amd.com/gpu in the backing store
        rc = backing.server.get("RayCluster", "ns1", "gc1")
        limits = rc["spec"]["workerGroupSpecs"][0]["template"]["spec"][
            "containers"][0]["resources"]["limits"]
        assert limits["amd.com/gpu"] == "2"

        got = self._call(channel, "ClusterService", "GetCluster",
                         GetRequest(name="gc1", namespace="ns1"), Cluster)
        assert got.version == "2.46.0"
        listed = self._call(channel, "ClusterService", "ListCluster",
                            ListRequest(namespace="ns1"), ListClusterResponse)
        assert len(listed.clusters) == 1
        self._call(channel, "ClusterService", "DeleteCluster",
                   DeleteRequest(name="gc1", namespace="ns1"), Empty)
        with pytest.raises(grpc.RpcError):
            self._call(channel, "ClusterService", "GetCluster",
                       GetRequest(name="gc1", namespace="ns1"), Cluster)


class TestGrpcFullSurface:
    """Conformance for the full six-service proto surface
    (cluster/config/job/job_submission/serve protos)."""

    @pytest.fixture()
    def grpc_stack(self):
        import grpc
        from kuberay_amd.apiserver.grpc_api import create_grpc_server
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.utils.fake_dashboard import FakeRayDashboardClient
        client = InMemoryClient()
        dashboard = FakeRayDashboardClient()
        server = create_grpc_server(client, port=0,
                                    dashboard_factory=lambda url: dashboard)
        port = server.add_insecure_port("127.0.0.1:0")
        server.start()
        channel = grpc.insecure_channel(f"127.0.0.1:{port}")
        yield channel, client, dashboard
        server.stop(0)

    def _call(self, channel, service, method, request, resp_cls):
        fn = channel.unary_unary(
            f"/kuberayamd.v1.{service}/{method}",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=resp_cls.FromString)
        return fn(request, timeout=5)

    def test_job_submission_service_round_trip(self, grpc_stack):
        """job_submission.proto:26-70: submit → details → log → list →
        stop → delete against a live cluster's dashboard."""
        import json as _json
        from kuberay_amd.apiserver.grpc_api import (
            Cluster, Empty, GetJobLogReply, JobSubmissionInfo,
            JobSubmissionQuery, ListJobSubmissionInfo, RayJobSubmission,
            SubmitRayJobReply, SubmitRayJobRequest)
        from kuberay_amd.testing import simple_raycluster
        channel, client, dashboard = grpc_stack
        client.create(simple_raycluster("subc", namespace="ns1"))
        reply = self._call(
            channel, "RayJobSubmissionService", "SubmitRayJob",
            SubmitRayJobRequest(
                namespace="ns1", clustername="subc",
                jobsubmission=RayJobSubmission(
                    entrypoint="python t.py",
                    metadata_json=_json.dumps({"team": "x"}))),
            SubmitRayJobReply)
        assert reply.submission_id
        details = self._call(
            channel, "RayJobSubmissionService", "GetJobDetails",
            JobSubmissionQuery(namespace="ns1", clustername="subc",
                               submissionid=reply.submission_id),
            JobSubmissionInfo)
        assert details.submission_id == reply.submission_id
        log = self._call(
            channel, "RayJobSubmissionService", "GetJobLog",
            JobSubmissionQuery(namespace="ns1", clustername="subc",
                               submissionid=reply.submission_id),
            GetJobLogReply)
        assert isinstance(log.log, str)
        listed = self._call(
            channel, "RayJobSubmissionService", "ListJobDetails",
            JobSubmissionQuery(namespace="ns1", clustername="subc"),
            ListJobSubmissionInfo)
        assert len(listed.submissions) == 1
        self._call(channel, "RayJobSubmissionService", "StopRayJob",
                   JobSubmissionQuery(namespace="ns1", clustername="subc",
                                      submissionid=reply.submission_id),
                   Empty)
        self._call(channel, "RayJobSubmissionService", "DeleteRayJob",
                   JobSubmissionQuery(namespace="ns1", clustername="subc",
                                      submissionid=reply.submission_id),
                   Empty)
        # removed from the dashboard's store (get_job_info auto-registers,
        # so assert on the raw map)
        assert reply.submission_id in dashboard.deleted_jobs
        assert reply.submission_id not in dashboard.jobs

    def test_job_submission_unknown_cluster_not_found(self, grpc_stack):
        import grpc as _grpc
        from kuberay_amd.apiserver.grpc_api import (
            JobSubmissionQuery, ListJobSubmissionInfo)
        channel, _, _ = grpc_stack
        with pytest.raises(_grpc.RpcError) as e:
            self._call(channel, "RayJobSubmissionService", "ListJobDetails",
                       JobSubmissionQuery(namespace="ns1",
                                          clustername="ghost"),
                       ListJobSubmissionInfo)
        assert e.value.code() == _grpc.StatusCode.NOT_FOUND

    def test_list_all_across_namespaces(self, grpc_stack):
        from kuberay_amd.apiserver.grpc_api import Empty, ListClusterResponse
        from kuberay_amd.testing import simple_raycluster
        channel, client, _ = grpc_stack
        client.create(simple_raycluster("a", namespace="ns1"))
        client.create(simple_raycluster("b", namespace="ns2"))
        listed = self._call(channel, "ClusterService", "ListAllClusters",
                            Empty(), ListClusterResponse)
        assert {c.namespace for c in listed.clusters} == {"ns1", "ns2"}

    def test_compute_template_get_and_list_all(self, grpc_stack):
        from kuberay_amd.apiserver.grpc_api import (
            ComputeTemplate, Empty, GetRequest, ListComputeTemplateResponse)
        channel, _, _ = grpc_stack
        for ns in ("ns1", "ns2"):
            self._call(channel, "ComputeTemplateService",
                       "CreateComputeTemplate",
                       ComputeTemplate(name=f"tpl-{ns}", namespace=ns,
                                       cpu=4, memory=8, gpu=1),
                       ComputeTemplate)
        got = self._call(channel, "ComputeTemplateService",
                         "GetComputeTemplate",
                         GetRequest(name="tpl-ns1", namespace="ns1"),
                         ComputeTemplate)
        assert got.gpu == 1 and got.gpu_accelerator == "amd.com/gpu"
        listed = self._call(channel, "ComputeTemplateService",
                            "ListAllComputeTemplates", Empty(),
                            ListComputeTemplateResponse)
        assert len(listed.compute_templates) == 2

    def test_image_template_crud(self, grpc_stack):
        import grpc as _grpc
        from kuberay_amd.apiserver.grpc_api import (
            DeleteRequest, Empty, GetRequest, ImageTemplate,
            ListImageTemplateResponse)
        channel, _, _ = grpc_stack
        tpl = ImageTemplate(name="rocm-base", namespace="ns1",
                            base_image="rocm/ray:2.46.0")
        tpl.pip_packages.append("numpy")
        self._call(channel, "ImageTemplateService", "CreateImageTemplate",
                   tpl, ImageTemplate)
        got = self._call(channel, "ImageTemplateService", "GetImageTemplate",
                         GetRequest(name="rocm-base", namespace="ns1"),
                         ImageTemplate)
        assert got.base_image == "rocm/ray:2.46.0"
        assert list(got.pip_packages) == ["numpy"]
        listed = self._call(channel, "ImageTemplateService",
                            "ListImageTemplates",
                            __import__("kuberay_amd.apiserver.grpc_api",
                                       fromlist=["ListRequest"])
                            .ListRequest(namespace="ns1"),
                            ListImageTemplateResponse)
        assert len(listed.image_templates) == 1
        self._call(channel, "ImageTemplateService", "DeleteImageTemplate",
                   DeleteRequest(name="rocm-base", namespace="ns1"), Empty)
        with pytest.raises(_grpc.RpcError):
            self._call(channel, "ImageTemplateService", "GetImageTemplate",
                       GetRequest(name="rocm-base", namespace="ns1"),
                       ImageTemplate)

    def test_update_ray_service(self, grpc_stack):
        from kuberay_amd.apiserver.grpc_api import RayServiceMsg
        channel, client, _ = grpc_stack
        import json as _json
        spec = {"headGroupSpec": {}, "workerGroupSpec": [
            {"groupName": "g", "replicas": 1, "maxReplicas": 2}]}
        self._call(channel, "RayServeService", "CreateRayService",
                   RayServiceMsg(name="svc", namespace="ns1",
                                 serve_config_v2="applications:\n- name: a\n",
                                 spec_json=_json.dumps(spec)), RayServiceMsg)
        updated = self._call(
            channel, "RayServeService", "UpdateRayService",
            RayServiceMsg(name="svc", namespace="ns1",
                          serve_config_v2="applications:\n- name: b\n"),
            RayServiceMsg)
        assert "name: b" in updated.serve_config_v2
        from kuberay_amd.models import RayService
        assert "name: b" in client.get(RayService, "ns1",
                                       "svc").spec.serve_config_v2

    def test_list_all_jobs_and_services(self, grpc_stack):
        import json as _json
        from kuberay_amd.apiserver.grpc_api import (
            Empty, ListRayJobResponse, ListRayServiceResponse, RayJobMsg)
        channel, client, _ = grpc_stack
        spec = {"clusterSpec": {"headGroupSpec": {}, "workerGroupSpec": []}}
        for ns in ("ns1", "ns2"):
            self._call(channel, "RayJobService", "CreateRayJob",
                       RayJobMsg(name=f"j-{ns}", namespace=ns,
                                 entrypoint="python t.py",
                                 spec_json=_json.dumps(spec)), RayJobMsg)
        listed = self._call(channel, "RayJobService", "ListAllRayJobs",
                            Empty(), ListRayJobResponse)
        assert {j.namespace for j in listed.jobs} == {"ns1", "ns2"}


class TestKraySuspend:
    def test_suspend_and_resume(self, kray):
        runner, backing = kray
        runner.invoke(cli, ["-n", "ns1", "create", "cluster", "k1"])
        r = runner.invoke(cli, ["-n", "ns1", "suspend", "cluster", "k1"])
        assert r.exit_code == 0, r.output
        assert backing.get(RayCluster, "ns1", "k1").spec.suspend is True
        r = runner.invoke(cli, ["-n", "ns1", "suspend", "cluster", "k1",
                                "--resume"])
        assert r.exit_code == 0
        assert backing.get(RayCluster, "ns1", "k1").spec.suspend is False


class TestServeApplicationsProxy:
    def test_put_and_get(self, api):
        t, _, fake = api
        t.post("/apis/v1/namespaces/ns1/clusters", json=CLUSTER_BODY)
        cfg = {"applications": [{"name": "app1", "import_path": "m.g"}]}
        r = t.put("/apis/v1/namespaces/ns1/serveapplications/c1", json=cfg)
        assert r.status_code == 200
        assert fake.serve_config == cfg
        r = t.get("/apis/v1/namespaces/ns1/serveapplications/c1")
        assert r.status_code == 200
        assert "applications" in r.json()


class TestKrayEvents:
    def test_get_events_lists_operator_events(self):
        import importlib
        from kuberay_amd.testing import ControlPlane, simple_raycluster
        climod = importlib.import_module("kuberay_amd.cli.main")
        cp = ControlPlane(kubelet_delay=0.01, poll_seconds=0.05).start()
        original = climod.make_client
        try:
            cp.client.create(simple_raycluster("demo"))
            assert cp.wait_cluster_state("default", "demo", "ready")
            climod.make_client = lambda server: cp.client
            r = CliRunner().invoke(cli, ["get", "events", "demo"])
            assert r.exit_code == 0, r.output
            assert "CreatedHeadPod" in r.output
        finally:
            climod.make_client = original
            cp.stop()


class TestV2ComputeTemplateMiddleware:
    def test_template_expanded_on_v2_create(self, api):
        t, client, _ = api
        body = {
            "apiVersion": "ray.io/v1", "kind": "RayCluster",
            "metadata": {"name": "mw1"},
            "spec": {
                "headGroupSpec": {"computeTemplate": "tpl",
                                  "rayStartParams": {}},
                "workerGroupSpecs": [{
                    "groupName": "g", "replicas": 1, "maxReplicas": 2,
                    "computeTemplate": "tpl", "rayStartParams": {}}],
            },
        }
        r = t.post("/apis/ray.io/v1/namespaces/ns1/rayclusters", json=body)
        assert r.status_code == 200, r.text
        rc = client.server.get("RayCluster", "ns1", "mw1")
        limits = rc["spec"]["workerGroupSpecs"][0]["template"]["spec"][
            "containers"][0]["resources"]["limits"]
        assert limits["amd.com/gpu"] == "2"

    def test_unknown_template_rejected(self, api):
        t, _, _ = api
        body = {"apiVersion": "ray.io/v1", "kind": "RayCluster",
                "metadata": {"name": "mw2"},
                "spec": {"headGroupSpec": {"computeTemplate": "missing"}}}
        r = t.post("/apis/ray.io/v1/namespaces/ns1/rayclusters", json=body)
        assert r.status_code == 400


class TestKrayNodeToken:
    def test_get_node_and_token(self):
        import importlib
        from kuberay_amd.testing import ControlPlane, simple_raycluster
        climod = importlib.import_module("kuberay_amd.cli.main")
        cp = ControlPlane(kubelet_delay=0.01, poll_seconds=0.05).start()
        original = climod.make_client
        try:
            climod.make_client = lambda server: cp.client
            cp.server.create({"kind": "Node", "metadata": {
                "name": "n1", "labels": {"amd.com/gpu.count": "8",
                                         "amd.com/xgmi-island": "n1-island0"}}})
            r = CliRunner().invoke(cli, ["get", "node"])
            assert r.exit_code == 0 and "n1-island0" in r.output
            cp.client.create(simple_raycluster(
                "authy", authOptions={"mode": "token"},
                rayVersion="2.53.0"))  # token auth needs Ray >= 2.52
            assert cp.wait_for(lambda: cp.server.try_get(
                "Secret", "default", "authy-auth-token"))
            r = CliRunner().invoke(cli, ["get", "token", "authy"])
            assert r.exit_code == 0 and len(r.output.strip()) == 64
        finally:
            climod.make_client = original
            cp.stop()


class TestPortForward:
    """kray session --forward: user-space TCP splice (kubectl-plugin
    session analog)."""

    def test_forward_round_trip(self):
        import socket
        import threading
        from kuberay_amd.cli.portforward import PortForwarder

        # a tiny upstream echo server standing in for the Ray dashboard
        upstream = socket.socket()
        upstream.bind(("127.0.0.1", 0))
        upstream.listen(4)
        uport = upstream.getsockname()[1]

        def echo_once():
            conn, _ = upstream.accept()
            data = conn.recv(1024)
            conn.sendall(b"echo:" + data)
            conn.close()

        threading.Thread(target=echo_once, daemon=True).start()

        fwd = PortForwarder("127.0.0.1", [(0, uport)]).start()
        try:
            local = fwd.local_ports[0]
            c = socket.create_connection(("127.0.0.1", local), timeout=5)
            c.sendall(b"ping")
            assert c.recv(1024) == b"echo:ping"
            c.close()
        finally:
            fwd.stop()
            upstream.close()

    def test_connection_refused_upstream_is_tolerated(self):
        import socket
        from kuberay_amd.cli.portforward import PortForwarder
        # remote port nobody listens on: local connect succeeds, then closes
        fwd = PortForwarder("127.0.0.1", [(0, 1)]).start()
        try:
            c = socket.create_connection(("127.0.0.1", fwd.local_ports[0]),
                                         timeout=5)
            c.settimeout(5)
            assert c.recv(1024) == b""  # closed cleanly, no hang
            c.close()
        finally:
            fwd.stop()

    def test_session_forward_requires_head_ip(self, kray):
        runner, client = kray
        from kuberay_amd.testing import simple_raycluster
        client.create(_ns(simple_raycluster("pfc1"), "ns1"))
        r = runner.invoke(cli, ["-n", "ns1", "session", "pfc1", "--forward"])
        assert r.exit_code != 0
        assert "no head pod IP" in r.output

    def test_session_forward_uses_head_ip(self, kray):
        import socket
        import threading
        runner, client = kray
        from kuberay_amd.models import RayCluster
        from kuberay_amd.testing import simple_raycluster
        client.create(_ns(simple_raycluster("pfc2"), "ns1"))
        rc = client.get(RayCluster, "ns1", "pfc2")
        rc.status.head.pod_ip = "127.0.0.1"
        client.update_status(rc)

        # upstream "dashboard" on an ephemeral port
        upstream = socket.socket()
        upstream.bind(("127.0.0.1", 0))
        upstream.listen(1)
        uport = upstream.getsockname()[1]

        def serve_once():
            conn, _ = upstream.accept()
            conn.sendall(b"dash")
            conn.close()

        threading.Thread(target=serve_once, daemon=True).start()

        # patch PortForwarder.wait so the command returns immediately,
        # keeping the forwarder alive for the assertion below
        from kuberay_amd.cli import portforward as pfmod
        held = {}
        orig_start = pfmod.PortForwarder.start

        def capture_start(self):
            held["fwd"] = self
            return orig_start(self)

        pfmod.PortForwarder.start = capture_start
        pfmod.PortForwarder.wait = lambda self: None
        try:
            r = runner.invoke(cli, ["-n", "ns1", "session", "pfc2",
                                    "--forward", "--port", f"0:{uport}"])
            assert r.exit_code == 0, r.output
            assert "forwarding 127.0.0.1:" in r.output
            local = held["fwd"].local_ports[0]
            c = socket.create_connection(("127.0.0.1", local), timeout=5)
            assert c.recv(1024) == b"dash"
            c.close()
        finally:
            held["fwd"].stop()
            upstream.close()
            pfmod.PortForwarder.start = orig_start
            del pfmod.PortForwarder.wait


def _ns(obj, namespace):
    obj.metadata.namespace = namespace
    return obj


class TestGrpcRayServeService:
    """serve.proto analog over the dynamic-proto gRPC server."""

    def test_service_crud(self):
        import grpc as grpclib
        from kuberay_amd.apiserver.grpc_api import (
            DeleteRequest, Empty, GetRequest, ListRequest,
            ListRayServiceResponse, RayServiceMsg, create_grpc_server)
        from kuberay_amd.kube.client import InMemoryClient
        from kuberay_amd.models import RayService
        client = InMemoryClient()
        server = create_grpc_server(client, port=0)
        port = server.add_insecure_port("127.0.0.1:0")
        server.start()
        try:
            channel = grpclib.insecure_channel(f"127.0.0.1:{port}")

            def call(method, request, resp_cls):
                fn = channel.unary_unary(
                    f"/kuberayamd.v1.RayServeService/{method}",
                    request_serializer=lambda m: m.SerializeToString(),
                    response_deserializer=resp_cls.FromString)
                return fn(request)

            import json as _json
            spec = {"rayClusterConfig": {
                "headGroupSpec": {"rayStartParams": {}, "template": {"spec": {
                    "containers": [{"name": "ray-head",
                                    "image": "rocm/ray:2.46.0"}]}}},
            }}
            created = call("CreateRayService", RayServiceMsg(
                name="s1", namespace="default",
                serve_config_v2="applications: []",
                spec_json=_json.dumps(spec)), RayServiceMsg)
            assert created.name == "s1"
            assert client.try_get(RayService, "default", "s1") is not None

            got = call("GetRayService", GetRequest(name="s1",
                                                   namespace="default"),
                       RayServiceMsg)
            assert got.serve_config_v2 == "applications: []"

            listed = call("ListRayServices", ListRequest(namespace="default"),
                          ListRayServiceResponse)
            assert [s.name for s in listed.services] == ["s1"]

            call("DeleteRayService", DeleteRequest(name="s1",
                                                   namespace="default"), Empty)
            assert client.try_get(RayService, "default", "s1") is None
        finally:
            server.stop(0)


class TestDashboardUi:
    """The multi-page dashboard shell (apiserver/dashboard.py): served at
    '/', contains the route set matching the reference's page tree
    (dashboard/src/app/{clusters,jobs,jobs/new,new}/page.tsx)."""

    def test_shell_served_with_all_pages(self, api):
        t, _, _ = api
        r = t.get("/")
        assert r.status_code == 200
        html = r.text
        for route in ("#/clusters", "#/jobs", "#/jobs/new", "#/services",
                      "#/new"):
            assert route in html, f"page route {route} missing from shell"
        # detail pages + the API paths the JS drives
        assert "pageClusterDetail" in html
        assert "pageServiceDetail" in html
        assert "/apis/v1/namespaces/" in html
        assert "/apis/ray.io/v1/namespaces/" in html

    def test_inline_gpu_cluster_create_shape(self, api):
        """The new-cluster form submits inline cpu/memory/gpu without a
        compute template; the converter must honor them."""
        t, client, _ = api
        r = t.post("/apis/v1/namespaces/ns1/clusters", json={
            "name": "ui-inline", "version": "2.46.0", "clusterSpec": {
                "headGroupSpec": {"rayStartParams": {}},
                "workerGroupSpec": [{"groupName": "mi355x-group",
                                     "replicas": 1, "minReplicas": 0,
                                     "maxReplicas": 8, "gpu": 4,
                                     "rayStartParams": {}}]}})
        assert r.status_code == 200, r.text
        rc = client.get(RayCluster, "ns1", "ui-inline")
        limits = rc.spec.worker_group_specs[0].template.spec.containers[0] \
            .resources.limits
        assert limits["amd.com/gpu"] == "4"


class TestDashboardFormContract:
    """Pins the exact request shapes the static dashboard's JS emits
    (createCluster/submitJob in apiserver/dashboard.py)."""

    def test_dashboard_cluster_create_shape(self, api):
        t, client, _ = api
        t.post("/apis/v1/namespaces/ns1/compute_templates",
               json={"name": "ui-tpl", "cpu": 4, "memory": 8, "gpu": 2})
        r = t.post("/apis/v1/namespaces/ns1/clusters", json={
            "name": "ui-c1", "version": "2.46.0", "clusterSpec": {
                "headGroupSpec": {"computeTemplate": "ui-tpl"},
                "workerGroupSpec": [{"groupName": "default-group",
                                     "computeTemplate": "ui-tpl",
                                     "replicas": 2, "minReplicas": 0,
                                     "maxReplicas": 8}]}})
        assert r.status_code == 200, r.text
        rc = client.get(RayCluster, "ns1", "ui-c1")
        limits = rc.spec.worker_group_specs[0].template.spec.containers[0] \
            .resources.limits
        assert limits["amd.com/gpu"] == "2"

    def test_dashboard_job_submit_shape(self, api):
        t, client, _ = api
        t.post("/apis/v1/namespaces/ns1/compute_templates",
               json={"name": "uj-tpl", "cpu": 4, "memory": 8, "gpu": 1})
        r = t.post("/apis/v1/namespaces/ns1/jobs", json={
            "name": "ui-j1", "entrypoint": "python train.py",
            "clusterSpec": {
                "headGroupSpec": {"computeTemplate": "uj-tpl"},
                "workerGroupSpec": [{"groupName": "default-group",
                                     "computeTemplate": "uj-tpl",
                                     "replicas": 1, "minReplicas": 0,
                                     "maxReplicas": 4}]}})
        assert r.status_code == 200, r.text
        from kuberay_amd.models import RayJob
        job = client.get(RayJob, "ns1", "ui-j1")
        assert job.spec.entrypoint == "python train.py"
        worker = job.spec.ray_cluster_spec.worker_group_specs[0]
        assert worker.template.spec.containers[0].resources.limits[
            "amd.com/gpu"] == "1"


def test_dashboard_service_create_shape(api):
    """Pins the RayService create request shape the dashboard JS emits."""
    t, client, _ = api
    t.post("/apis/v1/namespaces/ns1/compute_templates",
           json={"name": "us-tpl", "cpu": 4, "memory": 8, "gpu": 1})
    r = t.post("/apis/v1/namespaces/ns1/services", json={
        "name": "ui-s1",
        "serveConfig_V2": "applications:\n- name: a\n  import_path: m.g\n",
        "clusterSpec": {
            "headGroupSpec": {"computeTemplate": "us-tpl"},
            "workerGroupSpec": [{"groupName": "default-group",
                                 "computeTemplate": "us-tpl", "replicas": 1,
                                 "minReplicas": 0, "maxReplicas": 4}]}})
    assert r.status_code == 200, r.text
    from kuberay_amd.models import RayService
    svc = client.get(RayService, "ns1", "ui-s1")
    assert svc.spec.serve_config_v2.startswith("applications:")


def test_kray_get_cronjob(kray):
    runner, client = kray
    from kuberay_amd.models import RayCronJob
    from kuberay_amd.testing import simple_raycluster
    spec = {"schedule": "0 9 * * *", "timeZone": "America/New_York",
            "jobTemplate": {"entrypoint": "python t.py",
                            "rayClusterSpec":
                                simple_raycluster("x").spec.to_dict()}}
    client.create(RayCronJob.from_dict({
        "apiVersion": "ray.io/v1", "kind": "RayCronJob",
        "metadata": {"name": "nightly", "namespace": "ns1"}, "spec": spec}))
    r = runner.invoke(cli, ["-n", "ns1", "get", "cronjob"])
    assert r.exit_code == 0, r.output
    assert "nightly" in r.output and "America/New_York" in r.output


def test_compute_template_tolerations_applied(api):
    """cluster.go: compute-template tolerations land on generated pods."""
    t, client, _ = api
    t.post("/apis/v1/namespaces/ns1/compute_templates", json={
        "name": "tol-tpl", "cpu": 2, "memory": 4, "gpu": 1,
        "tolerations": [{"key": "amd.com/gpu", "operator": "Exists",
                         "effect": "NoSchedule"}]})
    r = t.post("/apis/v1/namespaces/ns1/clusters", json={
        "name": "tol-c1", "clusterSpec": {
            "headGroupSpec": {"computeTemplate": "tol-tpl"},
            "workerGroupSpec": [{"groupName": "g", "computeTemplate":
                                 "tol-tpl", "replicas": 1}]}})
    assert r.status_code == 200, r.text
    rc = client.get(RayCluster, "ns1", "tol-c1")
    for template in (rc.spec.head_group_spec.template,
                     rc.spec.worker_group_specs[0].template):
        tol = template.spec.to_dict().get("tolerations")
        assert tol and tol[0]["key"] == "amd.com/gpu"


class TestHttpClientRetry:
    """apiserversdk proxy.go:106-208 retry round-tripper analog."""

    def test_transient_5xx_retried_then_succeeds(self):
        import httpx as hx

        from kuberay_amd.kube.httpclient import HttpKubeClient
        calls = {"n": 0}

        def handler(request):
            calls["n"] += 1
            if calls["n"] < 3:
                return hx.Response(503, text="apiserver hiccup")
            return hx.Response(200, json={
                "apiVersion": "ray.io/v1", "kind": "RayCluster",
                "metadata": {"name": "r1", "namespace": "ns1"},
                "spec": {"headGroupSpec": {"rayStartParams": {},
                                           "template": {"spec": {
                                               "containers": []}}}}})

        client = HttpKubeClient(
            "http://x", http_client=hx.Client(
                transport=hx.MockTransport(handler), base_url="http://x"))
        rc = client.get(RayCluster, "ns1", "r1")
        assert rc.metadata.name == "r1"
        assert calls["n"] == 3

    def test_persistent_5xx_raises_after_retries(self):
        import httpx as hx

        from kuberay_amd.kube.httpclient import HttpKubeClient
        from kuberay_amd.kube.store import ApiError
        client = HttpKubeClient(
            "http://x", http_client=hx.Client(
                transport=hx.MockTransport(
                    lambda r: hx.Response(500, text="boom")),
                base_url="http://x"))
        with pytest.raises(ApiError):
            client.get(RayCluster, "ns1", "r1")

    def test_4xx_not_retried(self):
        import httpx as hx

        from kuberay_amd.kube.httpclient import HttpKubeClient
        from kuberay_amd.kube.store import NotFoundError
        calls = {"n": 0}

        def handler(request):
            calls["n"] += 1
            return hx.Response(404, text="nope")

        client = HttpKubeClient(
            "http://x", http_client=hx.Client(
                transport=hx.MockTransport(handler), base_url="http://x"))
        with pytest.raises(NotFoundError):
            client.get(RayCluster, "ns1", "r1")
        assert calls["n"] == 1


def test_kray_create_service(kray, tmp_path):
    runner, client = kray
    cfg = tmp_path / "serve.yaml"
    cfg.write_text("applications:\n- name: a\n  import_path: m.g\n")
    r = runner.invoke(cli, ["-n", "ns1", "create", "service", "svc-cli",
                            "--serve-config", str(cfg), "--worker-gpu", "2"])
    assert r.exit_code == 0, r.output
    from kuberay_amd.models import RayService
    svc = client.get(RayService, "ns1", "svc-cli")
    assert svc.spec.serve_config_v2.startswith("applications:")
    limits = svc.spec.ray_cluster_spec.worker_group_specs[0].template.spec \
        .containers[0].resources.limits
    assert limits["amd.com/gpu"] == "2"


def test_kray_create_cronjob(kray):
    runner, client = kray
    r = runner.invoke(cli, ["-n", "ns1", "create", "cronjob", "nightly2",
                            "--schedule", "0 3 * * *",
                            "--timezone", "America/New_York",
                            "--entrypoint", "python train.py",
                            "--worker-gpu", "1"])
    assert r.exit_code == 0, r.output
    from kuberay_amd.models import RayCronJob
    cron = client.get(RayCronJob, "ns1", "nightly2")
    assert cron.spec.schedule == "0 3 * * *"
    assert cron.spec.time_zone == "America/New_York"
    assert cron.spec.job_template.entrypoint == "python train.py"


class TestApiServerPythonClient:
    """python-apiserver-client analog: typed client over v1 endpoints."""

    def test_full_surface(self, api):
        t, backing, _ = api
        from kuberay_amd.apiserver.client import (ApiServerClient,
                                                  ApiServerError)
        c = ApiServerClient(http_client=t)
        c.create_compute_template("ns1", {"name": "pyc-tpl", "cpu": 2,
                                          "memory": 4, "gpu": 1})
        assert any(x["name"] == "pyc-tpl"
                   for x in c.list_compute_templates("ns1"))
        spec = {"headGroupSpec": {"computeTemplate": "pyc-tpl"},
                "workerGroupSpec": [{"groupName": "g",
                                     "computeTemplate": "pyc-tpl",
                                     "replicas": 1}]}
        c.create_cluster("ns1", {"name": "pyc-c1", "clusterSpec": spec})
        assert c.get_cluster("ns1", "pyc-c1")["name"] == "pyc-c1"
        c.create_job("ns1", {"name": "pyc-j1", "entrypoint": "python x.py",
                             "clusterSpec": spec})
        assert [j["name"] for j in c.list_jobs("ns1")] == ["pyc-j1"]
        c.create_service("ns1", {"name": "pyc-s1",
                                 "serveConfig_V2": "applications: []",
                                 "clusterSpec": spec})
        assert c.get_service("ns1", "pyc-s1")["name"] == "pyc-s1"
        for deleter, name in ((c.delete_service, "pyc-s1"),
                              (c.delete_job, "pyc-j1"),
                              (c.delete_cluster, "pyc-c1"),
                              (c.delete_compute_template, "pyc-tpl")):
            deleter("ns1", name)
        with pytest.raises(ApiServerError):
            c.get_cluster("ns1", "pyc-c1")


class TestKrayOutputFormats:
    """kubectl -o yaml|json parity for kray get."""

    def test_get_cluster_yaml_and_json(self, kray):
        import json as _json

        import yaml as _yaml
        runner, backing = kray
        runner.invoke(cli, ["-n", "ns1", "create", "cluster", "k1",
                            "--worker-gpu", "2"])
        r = runner.invoke(cli, ["-n", "ns1", "get", "cluster", "k1",
                                "-o", "yaml"])
        assert r.exit_code == 0, r.output
        doc = _yaml.safe_load(r.output)
        assert doc["kind"] == "RayCluster"
        assert doc["spec"]["workerGroupSpecs"][0]["template"]["spec"][
            "containers"][0]["resources"]["limits"]["amd.com/gpu"] == "2"
        r = runner.invoke(cli, ["-n", "ns1", "get", "cluster", "k1",
                                "-o", "json"])
        assert r.exit_code == 0
        assert _json.loads(r.output)["metadata"]["name"] == "k1"

    def test_get_cluster_list_json_envelope(self, kray):
        import json as _json
        runner, _ = kray
        runner.invoke(cli, ["-n", "ns1", "create", "cluster", "a1"])
        runner.invoke(cli, ["-n", "ns1", "create", "cluster", "a2"])
        r = runner.invoke(cli, ["-n", "ns1", "get", "cluster", "-o", "json"])
        out = _json.loads(r.output)
        assert out["kind"] == "List" and len(out["items"]) == 2

    def test_bad_output_format_rejected(self, kray):
        runner, _ = kray
        r = runner.invoke(cli, ["-n", "ns1", "get", "cluster", "-o", "wide"])
        assert r.exit_code != 0
        assert "unsupported output format" in r.output


class TestV2ComputeTemplateNested:
    """Round-2 extension of the v2 compute-template middleware: RayJob
    (spec.rayClusterSpec) and RayService (spec.rayClusterConfig) shapes are
    expanded too (apiserversdk/util/template.go:54-79)."""

    def test_rayjob_nested_expansion(self, api):
        t, _, _ = api
        r = t.post("/apis/ray.io/v1/namespaces/ns1/rayjobs", json={
            "apiVersion": "ray.io/v1", "kind": "RayJob",
            "metadata": {"name": "v2job"},
            "spec": {"entrypoint": "python x.py",
                     "rayClusterSpec": {
                         "headGroupSpec": {"computeTemplate": "tpl",
                                           "rayStartParams": {}},
                         "workerGroupSpecs": [{
                             "groupName": "g", "replicas": 1,
                             "maxReplicas": 2, "computeTemplate": "tpl",
                             "rayStartParams": {}}]}}})
        assert r.status_code == 200, r.text
        got = t.get("/apis/ray.io/v1/namespaces/ns1/rayjobs/v2job").json()
        limits = got["spec"]["rayClusterSpec"]["workerGroupSpecs"][0][
            "template"]["spec"]["containers"][0]["resources"]["limits"]
        assert limits["amd.com/gpu"] == "2"

    def test_rayservice_nested_expansion(self, api):
        t, _, _ = api
        r = t.post("/apis/ray.io/v1/namespaces/ns1/rayservices", json={
            "apiVersion": "ray.io/v1", "kind": "RayService",
            "metadata": {"name": "v2svc"},
            "spec": {"serveConfigV2": "applications:\n- name: a\n",
                     "rayClusterConfig": {
                         "headGroupSpec": {"computeTemplate": "tpl",
                                           "rayStartParams": {}},
                         "workerGroupSpecs": [{
                             "groupName": "g", "replicas": 1,
                             "maxReplicas": 2, "computeTemplate": "tpl",
                             "rayStartParams": {}}]}}})
        assert r.status_code == 200, r.text
        got = t.get("/apis/ray.io/v1/namespaces/ns1/rayservices/v2svc").json()
        limits = got["spec"]["rayClusterConfig"]["workerGroupSpecs"][0][
            "template"]["spec"]["containers"][0]["resources"]["limits"]
        assert limits["amd.com/gpu"] == "2"


class TestImageTemplatesHttp:
    """HTTP mapping of ImageTemplateService (grpc-gateway parity)."""

    def test_crud(self, api):
        t, _, _ = api
        r = t.post("/apis/v1/namespaces/ns1/image_templates", json={
            "name": "rocm-base", "baseImage": "rocm/ray:2.46.0",
            "pipPackages": ["numpy", "scipy"],
            "environmentVariables": {"HSA_ENABLE_IPC_MODE_LEGACY": "0"}})
        assert r.status_code == 200
        got = t.get(
            "/apis/v1/namespaces/ns1/image_templates/rocm-base").json()
        assert got["baseImage"] == "rocm/ray:2.46.0"
        assert got["pipPackages"] == ["numpy", "scipy"]
        listed = t.get("/apis/v1/namespaces/ns1/image_templates").json()
        assert len(listed["imageTemplates"]) == 1
        assert t.delete(
            "/apis/v1/namespaces/ns1/image_templates/rocm-base"
        ).status_code == 200
        assert t.get(
            "/apis/v1/namespaces/ns1/image_templates/rocm-base"
        ).status_code == 404

    def test_grpc_and_http_share_storage(self, api):
        """The gRPC ImageTemplateService and the HTTP routes read/write the
        same ConfigMaps."""
        t, client, _ = api
        t.post("/apis/v1/namespaces/ns1/image_templates", json={
            "name": "shared", "baseImage": "img"})
        import grpc as _grpc
        from kuberay_amd.apiserver.grpc_api import (
            GetRequest, ImageTemplate, create_grpc_server)
        server = create_grpc_server(client, port=0)
        port = server.add_insecure_port("127.0.0.1:0")
        server.start()
        try:
            channel = _grpc.insecure_channel(f"127.0.0.1:{port}")
            fn = channel.unary_unary(
                "/kuberayamd.v1.ImageTemplateService/GetImageTemplate",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=ImageTemplate.FromString)
            got = fn(GetRequest(name="shared", namespace="ns1"), timeout=5)
            assert got.base_image == "img"
        finally:
            server.stop(0)


def test_list_all_routes_cross_namespace(api):
    """grpc-gateway ListAll* HTTP routes return objects from every
    namespace (cluster.proto:50 / job.proto:52 / service.proto:49 /
    config.proto:50)."""
    t, _, _ = api
    for ns in ("nsa", "nsb"):
        r = t.post(f"/apis/v1/namespaces/{ns}/clusters", json={
            "name": f"c-{ns}", "clusterSpec": {
                "headGroupSpec": {"computeTemplate": "", "image": "i",
                                  "rayStartParams": {}},
                "workerGroupSpec": []}})
        assert r.status_code == 200, r.text
    names = {c["name"] for c in t.get("/apis/v1/clusters").json()["clusters"]}
    assert {"c-nsa", "c-nsb"} <= names
    assert t.get("/apis/v1/jobs").json()["jobs"] == []
    assert t.get("/apis/v1/services").json()["services"] == []
    assert "computeTemplates" in t.get("/apis/v1/compute_templates").json()
