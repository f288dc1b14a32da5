"""Native C++ store backend: parity with the python backend + view tests."""
import pytest

from kuberay_amd.kube.store import InMemoryApiServer, PyBackend, compute_pod_view

try:
    from kuberay_amd.kube.native import NativeBackend
    HAVE_NATIVE = True
except ImportError:
    HAVE_NATIVE = False

pytestmark = pytest.mark.skipif(not HAVE_NATIVE, reason="native engine not built")


POD = {
    "apiVersion": "v1", "kind": "Pod",
    "metadata": {"name": "p1", "namespace": "ns1",
                 "labels": {"ray.io/cluster": "c1", "ray.io/node-type": "worker"}},
    "spec": {"containers": [{"name": "ray"}], "restartPolicy": "Never"},
    "status": {"phase": "Running", "podIP": "10.0.0.9",
               "conditions": [{"type": "Ready", "status": "True"}]},
}


@pytest.fixture(params=["python", "native"])
def server(request):
    backend = PyBackend() if request.param == "python" else NativeBackend()
    return InMemoryApiServer(backend=backend)


class TestBackendParity:
    def test_crud_roundtrip(self, server):
        created = server.create(dict(POD))
        assert created["metadata"]["uid"]
        got = server.get("Pod", "ns1", "p1")
        assert got["status"]["podIP"] == "10.0.0.9"
        got["spec"]["restartPolicy"] = "Always"
        updated = server.update(got)
        assert updated["metadata"]["generation"] == 2
        server.delete("Pod", "ns1", "p1")
        assert server.try_get("Pod", "ns1", "p1") is None

    def test_label_list(self, server):
        server.create(dict(POD))
        pod2 = {"kind": "Pod", "metadata": {
            "name": "p2", "namespace": "ns1",
            "labels": {"ray.io/cluster": "c2"}}}
        server.create(pod2)
        out = server.list("Pod", "ns1", {"ray.io/cluster": "c1"})
        assert [o["metadata"]["name"] for o in out] == ["p1"]
        assert server.count("Pod") == 2

    def test_views(self, server):
        server.create(dict(POD))
        views = server.list_pod_views("ns1", {"ray.io/cluster": "c1"})
        assert len(views) == 1
        v = views[0]
        assert v.name == "p1" and v.phase == "Running" and v.ready
        assert v.pod_ip == "10.0.0.9" and v.restart_policy == "Never"
        assert v.labels["ray.io/node-type"] == "worker"

    def test_view_updates_with_status(self, server):
        server.create(dict(POD))
        server.patch_merge("Pod", "ns1", "p1", {
            "status": {"phase": "Failed"}}, subresource="status")
        v = server.list_pod_views("ns1")[0]
        assert v.phase == "Failed"

    def test_owner_gc(self, server):
        owner = server.create({"kind": "RayCluster",
                               "metadata": {"name": "c1", "namespace": "ns1"}})
        dep = dict(POD)
        dep["metadata"] = dict(dep["metadata"])
        dep["metadata"]["ownerReferences"] = [{
            "kind": "RayCluster", "name": "c1", "uid": owner["metadata"]["uid"]}]
        server.create(dep)
        server.delete("RayCluster", "ns1", "c1")
        assert server.try_get("Pod", "ns1", "p1") is None

    def test_terminated_container_flag(self, server):
        pod = dict(POD)
        pod["status"] = {
            "phase": "Running",
            "containerStatuses": [{"name": "ray",
                                   "state": {"terminated": {"exitCode": 1}}}]}
        server.create(pod)
        v = server.list_pod_views("ns1")[0]
        assert v.ray_container_terminated


class TestNativeSpecifics:
    def test_default_backend_is_native(self):
        s = InMemoryApiServer()
        assert s.backend_name == "native-cpp"

    def test_blob_accounting(self):
        b = NativeBackend()
        b.put(("Pod", "ns1", "p1"), dict(POD))
        assert 0 < b.total_bytes() < 2048
