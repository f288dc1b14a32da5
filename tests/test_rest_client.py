"""REST client tests against a mocked K8s apiserver (httpx MockTransport)."""
import base64
import json

import httpx
import pytest

from kuberay_amd.kube.rest import RESOURCES, RestClient, _load_kubeconfig
from kuberay_amd.kube.store import AlreadyExistsError, ConflictError, NotFoundError
from kuberay_amd.models import RayCluster
from kuberay_amd.testing import simple_raycluster


class FakeK8s:
    """Tiny in-memory K8s REST handler behind MockTransport."""

    def __init__(self):
        self.objects = {}

    def handler(self, request: httpx.Request) -> httpx.Response:
        parts = request.url.path.strip("/").split("/")
        # .../namespaces/{ns}/{plural}[/{name}[/status]]
        ns_idx = parts.index("namespaces") if "namespaces" in parts else -1
        ns = parts[ns_idx + 1] if ns_idx >= 0 else None
        plural = parts[ns_idx + 2] if ns_idx >= 0 else parts[-1]
        name = parts[ns_idx + 3] if ns_idx >= 0 and len(parts) > ns_idx + 3 else None
        if name == "status":
            name = None
        sub = parts[-1] if parts[-1] == "status" else None

        key = (plural, ns, name)
        if request.method == "POST":
            body = json.loads(request.content)
            k = (plural, ns, body["metadata"]["name"])
            if k in self.objects:
                return httpx.Response(409, text='AlreadyExists')
            body["metadata"]["uid"] = "uid-1"
            body["metadata"]["resourceVersion"] = "1"
            self.objects[k] = body
            return httpx.Response(200, json=body)
        if request.method == "GET" and name:
            obj = self.objects.get(key)
            if obj is None:
                return httpx.Response(404, text="not found")
            return httpx.Response(200, json=obj)
        if request.method == "GET":
            sel = request.url.params.get("labelSelector")
            items = [o for (p, n, _), o in self.objects.items()
                     if p == plural and (ns is None or n == ns)]
            if sel:
                want = dict(kv.split("=") for kv in sel.split(","))
                items = [o for o in items
                         if all((o["metadata"].get("labels") or {}).get(k) == v
                                for k, v in want.items())]
            return httpx.Response(200, json={"items": items})
        if request.method == "PUT":
            body = json.loads(request.content)
            k = (plural, ns, body["metadata"]["name"])
            if k not in self.objects:
                return httpx.Response(404, text="not found")
            self.objects[k] = body
            return httpx.Response(200, json=body)
        if request.method == "PATCH":
            obj = self.objects.get(key)
            if obj is None:
                return httpx.Response(404, text="not found")
            patch = json.loads(request.content)

            def merge(dst, src):
                for k2, v in src.items():
                    if isinstance(v, dict) and isinstance(dst.get(k2), dict):
                        merge(dst[k2], v)
                    else:
                        dst[k2] = v
            merge(obj, patch)
            return httpx.Response(200, json=obj)
        if request.method == "DELETE":
            if key not in self.objects:
                return httpx.Response(404, text="not found")
            del self.objects[key]
            return httpx.Response(200, json={"status": "Success"})
        return httpx.Response(400, text="bad request")


@pytest.fixture()
def rest():
    fake = FakeK8s()
    http = httpx.Client(base_url="https://k8s.test",
                        transport=httpx.MockTransport(fake.handler))
    return RestClient(http_client=http), fake


class TestRestClient:
    def test_create_get_update_delete(self, rest):
        client, fake = rest
        client.create(simple_raycluster("r1", namespace="ns1"))
        got = client.get(RayCluster, "ns1", "r1")
        assert got.metadata.uid == "uid-1"
        got.spec.worker_group_specs[0].replicas = 7
        client.update(got)
        assert fake.objects[("rayclusters", "ns1", "r1")]["spec"][
            "workerGroupSpecs"][0]["replicas"] == 7
        client.delete(RayCluster, "ns1", "r1")
        with pytest.raises(NotFoundError):
            client.get(RayCluster, "ns1", "r1")

    def test_duplicate_create_raises(self, rest):
        client, _ = rest
        client.create(simple_raycluster("r1", namespace="ns1"))
        with pytest.raises(AlreadyExistsError):
            client.create(simple_raycluster("r1", namespace="ns1"))

    def test_label_selector_list(self, rest):
        client, _ = rest
        a = simple_raycluster("a", namespace="ns1")
        a.metadata.labels = {"team": "x"}
        b = simple_raycluster("b", namespace="ns1")
        client.create(a)
        client.create(b)
        out = client.list(RayCluster, "ns1", {"team": "x"})
        assert [o.metadata.name for o in out] == ["a"]

    def test_patch_merge(self, rest):
        client, fake = rest
        client.create(simple_raycluster("r1", namespace="ns1"))
        client.patch(RayCluster, "ns1", "r1",
                     {"metadata": {"labels": {"x": "y"}}})
        assert fake.objects[("rayclusters", "ns1", "r1")]["metadata"][
            "labels"]["x"] == "y"

    def test_resource_paths_cover_all_owned_kinds(self):
        for kind in ("Pod", "Service", "Secret", "Job", "PersistentVolumeClaim",
                     "RayCluster", "RayJob", "RayService", "RayCronJob",
                     "NetworkPolicy", "Role", "RoleBinding"):
            assert kind in RESOURCES


class TestKubeconfig:
    def test_token_auth(self, tmp_path):
        cfg = {
            "current-context": "c",
            "contexts": [{"name": "c", "context": {"cluster": "cl", "user": "u"}}],
            "clusters": [{"name": "cl", "cluster": {
                "server": "https://1.2.3.4:6443",
                "insecure-skip-tls-verify": True}}],
            "users": [{"name": "u", "user": {"token": "tok123"}}],
        }
        import yaml
        p = tmp_path / "kubeconfig"
        p.write_text(yaml.safe_dump(cfg))
        server, kwargs = _load_kubeconfig(str(p))
        assert server == "https://1.2.3.4:6443"
        assert kwargs["headers"]["Authorization"] == "Bearer tok123"
        assert kwargs["verify"] is False

    def test_ca_data(self, tmp_path):
        import yaml
        cfg = {
            "current-context": "c",
            "contexts": [{"name": "c", "context": {"cluster": "cl", "user": "u"}}],
            "clusters": [{"name": "cl", "cluster": {
                "server": "https://h:6443",
                "certificate-authority-data":
                    base64.b64encode(b"CERT").decode()}}],
            "users": [{"name": "u", "user": {"token": "t"}}],
        }
        p = tmp_path / "kubeconfig"
        p.write_text(yaml.safe_dump(cfg))
        server, kwargs = _load_kubeconfig(str(p))
        assert isinstance(kwargs["verify"], str)
        assert open(kwargs["verify"], "rb").read() == b"CERT"


def test_pod_logs_via_rest():
    """kubectl-plugin log-download analog: GET pods/{name}/log."""
    import httpx
    from kuberay_amd.kube.rest import RestClient

    def handler(request):
        if request.url.path.endswith("/pods/p1/log"):
            assert request.url.params["tailLines"] == "5"
            return httpx.Response(200, text="line1\nline2\n")
        return httpx.Response(404, text="not found")

    http = httpx.Client(transport=httpx.MockTransport(handler),
                        base_url="http://kube")
    client = RestClient(http_client=http)
    assert client.pod_logs("ns1", "p1", tail_lines=5) == "line1\nline2\n"
