"""mTLS controller tests (reference analog: raycluster_mtls_controller_test.go)."""
import base64

import pytest

from kuberay_amd.kube.client import InMemoryClient
from kuberay_amd.kube import objects as k8s
from kuberay_amd.models import RayCluster
from kuberay_amd.ops.mtls import (
    MTLSReconciler,
    cert_sans,
    generate_ca,
    generate_leaf,
)
from kuberay_amd.testing import simple_raycluster


class TestCertGeneration:
    def test_ca_and_leaf(self):
        ca_cert, ca_key = generate_ca("test-ca")
        cert, key = generate_leaf(ca_cert, ca_key, "head",
                                  ["svc.ns.svc.cluster.local", "localhost"],
                                  ["127.0.0.1", "10.0.0.5"])
        assert b"BEGIN CERTIFICATE" in cert
        assert b"BEGIN PRIVATE KEY" in key
        sans = cert_sans(cert)
        assert "svc.ns.svc.cluster.local" in sans
        assert "10.0.0.5" in sans


class TestSelfSignedReconciler:
    def _cluster(self):
        return simple_raycluster("demo", tlsOptions={"enabled": True})

    def test_secrets_issued(self):
        client = InMemoryClient()
        client.create(self._cluster())
        r = MTLSReconciler(client)
        r.reconcile(("default", "demo"))
        for name in ("ca-secret-demo", "ray-head-secret-demo",
                     "ray-worker-secret-demo"):
            secret = client.try_get(k8s.Secret, "default", name)
            assert secret is not None, name
        head = client.get(k8s.Secret, "default", "ray-head-secret-demo")
        assert set(head.data) == {"tls.crt", "tls.key", "ca.crt"}
        cert = base64.b64decode(head.data["tls.crt"])
        assert "demo-head-svc" in cert_sans(cert)

    def test_head_cert_reissued_when_pod_ip_appears(self):
        client = InMemoryClient()
        client.create(self._cluster())
        r = MTLSReconciler(client)
        r.reconcile(("default", "demo"))
        # a head pod shows up with an IP not in the SANs
        client.server.create({
            "kind": "Pod",
            "metadata": {"name": "demo-head-x", "namespace": "default",
                         "labels": {"ray.io/cluster": "demo",
                                    "ray.io/node-type": "head"}},
            "spec": {"containers": [{"name": "ray"}]},
            "status": {"phase": "Running", "podIP": "10.99.1.2"}})
        r.reconcile(("default", "demo"))
        head = client.get(k8s.Secret, "default", "ray-head-secret-demo")
        assert "10.99.1.2" in cert_sans(base64.b64decode(head.data["tls.crt"]))

    def _add_pod(self, client, name, node_type, ip):
        client.server.create({
            "kind": "Pod",
            "metadata": {"name": name, "namespace": "default",
                         "labels": {"ray.io/cluster": "demo",
                                    "ray.io/node-type": node_type}},
            "spec": {"containers": [{"name": "ray"}]},
            "status": {"phase": "Running", "podIP": ip}})

    def test_worker_cert_tracks_worker_pod_ips(self):
        """GCS dials workers by pod IP — the WORKER cert must carry worker
        pod IPs (reference reconcileWorkerCertificate)."""
        client = InMemoryClient()
        client.create(self._cluster())
        r = MTLSReconciler(client)
        r.reconcile(("default", "demo"))
        self._add_pod(client, "demo-worker-a", "worker", "10.99.2.7")
        r.reconcile(("default", "demo"))
        worker = client.get(k8s.Secret, "default", "ray-worker-secret-demo")
        assert "10.99.2.7" in cert_sans(
            base64.b64decode(worker.data["tls.crt"]))

    def test_head_cert_not_reissued_on_worker_scale(self):
        """Worker churn must not force a head cert reissue: the head cert
        carries head pod IPs only (reference reconcileHeadCertificate)."""
        client = InMemoryClient()
        client.create(self._cluster())
        r = MTLSReconciler(client)
        self._add_pod(client, "demo-head-x", "head", "10.99.1.2")
        r.reconcile(("default", "demo"))
        head1 = client.get(k8s.Secret, "default", "ray-head-secret-demo")
        sans1 = cert_sans(base64.b64decode(head1.data["tls.crt"]))
        assert "10.99.1.2" in sans1
        # worker scale event
        self._add_pod(client, "demo-worker-a", "worker", "10.99.2.7")
        r.reconcile(("default", "demo"))
        head2 = client.get(k8s.Secret, "default", "ray-head-secret-demo")
        assert head2.metadata.resource_version == head1.metadata.resource_version
        assert "10.99.2.7" not in cert_sans(
            base64.b64decode(head2.data["tls.crt"]))

    def test_sans_cover_is_exact_not_substring(self):
        """10.0.0.11 in the cert must NOT 'cover' a required 10.0.0.1."""
        client = InMemoryClient()
        client.create(self._cluster())
        r = MTLSReconciler(client)
        self._add_pod(client, "demo-head-x", "head", "10.0.0.11")
        r.reconcile(("default", "demo"))
        head1 = client.get(k8s.Secret, "default", "ray-head-secret-demo")
        assert "10.0.0.11" in cert_sans(base64.b64decode(head1.data["tls.crt"]))
        # second head IP that is a prefix of the existing SAN entry
        self._add_pod(client, "demo-head-y", "head", "10.0.0.1")
        r.reconcile(("default", "demo"))
        head2 = client.get(k8s.Secret, "default", "ray-head-secret-demo")
        sans = cert_sans(base64.b64decode(head2.data["tls.crt"]))
        assert "IP Address:10.0.0.1," in sans + "," or \
            sans.rstrip().endswith("10.0.0.1")
        assert head2.metadata.resource_version != head1.metadata.resource_version

    def test_noop_without_tls(self):
        client = InMemoryClient()
        client.create(simple_raycluster("demo"))
        MTLSReconciler(client).reconcile(("default", "demo"))
        assert client.server.count("Secret") == 0

    def test_stable_when_sans_covered(self):
        client = InMemoryClient()
        client.create(self._cluster())
        r = MTLSReconciler(client)
        r.reconcile(("default", "demo"))
        rv1 = client.get(k8s.Secret, "default", "ray-head-secret-demo") \
            .metadata.resource_version
        r.reconcile(("default", "demo"))
        rv2 = client.get(k8s.Secret, "default", "ray-head-secret-demo") \
            .metadata.resource_version
        assert rv1 == rv2


class TestCertManagerMode:
    def test_creates_cert_manager_objects(self):
        client = InMemoryClient()
        client.create(simple_raycluster("demo", tlsOptions={"enabled": True}))
        r = MTLSReconciler(client, mode="cert-manager")
        r.reconcile(("default", "demo"))
        issuers = client.server.list("Issuer")
        certs = client.server.list("Certificate")
        assert len(issuers) == 2 and len(certs) == 3
        ca_cert = next(c for c in certs
                       if c["metadata"]["name"] == "ray-ca-certificate-demo")
        assert ca_cert["spec"]["isCA"] is True

    def test_certificate_ip_sans_track_per_role_pods(self):
        client = InMemoryClient()
        client.create(simple_raycluster("demo", tlsOptions={"enabled": True}))
        r = MTLSReconciler(client, mode="cert-manager")
        r.reconcile(("default", "demo"))
        for name, node_type, ip in (("demo-head-x", "head", "10.99.1.2"),
                                    ("demo-worker-a", "worker", "10.99.2.7")):
            client.server.create({
                "kind": "Pod",
                "metadata": {"name": name, "namespace": "default",
                             "labels": {"ray.io/cluster": "demo",
                                        "ray.io/node-type": node_type}},
                "spec": {"containers": [{"name": "ray"}]},
                "status": {"phase": "Running", "podIP": ip}})
        r.reconcile(("default", "demo"))
        head = client.server.get("Certificate", "default", "ray-head-cert-demo")
        worker = client.server.get("Certificate", "default",
                                   "ray-worker-cert-demo")
        assert head["spec"]["ipAddresses"] == ["10.99.1.2", "127.0.0.1"]
        assert worker["spec"]["ipAddresses"] == ["10.99.2.7", "127.0.0.1"]


class TestRealTlsHandshake:
    """Crypto-level e2e (the sim can't run real pods, but the certs are
    real): a server presenting the operator-issued HEAD cert and a client
    trusting only the operator CA complete a mutual-TLS handshake with
    hostname verification against the issued SANs."""

    def test_mutual_handshake_with_issued_certs(self, tmp_path):
        import socket
        import ssl
        import threading

        client = InMemoryClient()
        client.create(simple_raycluster("demo", tlsOptions={"enabled": True}))
        client.server.create({
            "kind": "Pod",
            "metadata": {"name": "demo-head-x", "namespace": "default",
                         "labels": {"ray.io/cluster": "demo",
                                    "ray.io/node-type": "head"}},
            "spec": {"containers": [{"name": "ray"}]},
            "status": {"phase": "Running", "podIP": "127.0.0.1"}})
        MTLSReconciler(client).reconcile(("default", "demo"))

        def write(name, secret_name, key):
            data = client.get(k8s.Secret, "default", secret_name).data[key]
            p = tmp_path / name
            p.write_bytes(base64.b64decode(data))
            return str(p)

        ca = write("ca.crt", "ca-secret-demo", "ca.crt")
        head_crt = write("head.crt", "ray-head-secret-demo", "tls.crt")
        head_key = write("head.key", "ray-head-secret-demo", "tls.key")
        worker_crt = write("worker.crt", "ray-worker-secret-demo", "tls.crt")
        worker_key = write("worker.key", "ray-worker-secret-demo", "tls.key")

        server_ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        server_ctx.load_cert_chain(head_crt, head_key)
        server_ctx.load_verify_locations(ca)
        server_ctx.verify_mode = ssl.CERT_REQUIRED  # mutual TLS

        client_ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
        client_ctx.load_verify_locations(ca)
        client_ctx.load_cert_chain(worker_crt, worker_key)
        client_ctx.check_hostname = True

        listener = socket.socket()
        listener.bind(("127.0.0.1", 0))
        listener.listen(1)
        port = listener.getsockname()[1]
        server_err = []

        def serve():
            try:
                conn, _ = listener.accept()
                with server_ctx.wrap_socket(conn, server_side=True) as tls:
                    assert tls.recv(5) == b"hello"
                    tls.sendall(b"world")
            except Exception as e:  # noqa: BLE001
                server_err.append(e)

        t = threading.Thread(target=serve, daemon=True)
        t.start()
        raw = socket.create_connection(("127.0.0.1", port), timeout=5)
        # hostname verification against the head cert's IP SAN (127.0.0.1)
        with client_ctx.wrap_socket(raw, server_hostname="127.0.0.1") as tls:
            tls.sendall(b"hello")
            assert tls.recv(5) == b"world"
        t.join(timeout=5)
        listener.close()
        assert not server_err, server_err

    def test_foreign_ca_rejected(self, tmp_path):
        """A cert from a DIFFERENT operator CA must fail verification —
        the trust boundary is per-cluster."""
        import socket
        import ssl
        import threading

        other_ca, other_key = generate_ca("other-ca")
        crt, key = generate_leaf(other_ca, other_key, "imposter",
                                 ["localhost"], ["127.0.0.1"])
        (tmp_path / "i.crt").write_bytes(crt)
        (tmp_path / "i.key").write_bytes(key)

        client = InMemoryClient()
        client.create(simple_raycluster("demo", tlsOptions={"enabled": True}))
        MTLSReconciler(client).reconcile(("default", "demo"))
        ca_pem = base64.b64decode(
            client.get(k8s.Secret, "default", "ca-secret-demo").data["ca.crt"])
        (tmp_path / "ca.crt").write_bytes(ca_pem)

        server_ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        server_ctx.load_cert_chain(str(tmp_path / "i.crt"),
                                   str(tmp_path / "i.key"))
        client_ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
        client_ctx.load_verify_locations(str(tmp_path / "ca.crt"))

        listener = socket.socket()
        listener.bind(("127.0.0.1", 0))
        listener.listen(1)
        port = listener.getsockname()[1]

        def serve():
            try:
                conn, _ = listener.accept()
                with server_ctx.wrap_socket(conn, server_side=True):
                    pass
            except Exception:  # noqa: BLE001 — expected: client aborts
                pass

        t = threading.Thread(target=serve, daemon=True)
        t.start()
        raw = socket.create_connection(("127.0.0.1", port), timeout=5)
        with pytest.raises(ssl.SSLError):
            client_ctx.wrap_socket(raw, server_hostname="127.0.0.1")
        raw.close()
        t.join(timeout=5)
        listener.close()
