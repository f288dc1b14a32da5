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
