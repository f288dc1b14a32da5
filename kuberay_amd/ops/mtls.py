"""mTLS controller (reference: raycluster_mtls_controller.go:66-593,
feature gate RayClusterMTLS).

The reference drives cert-manager (self-signed Issuer → CA Certificate →
CA Issuer → head/worker leaf Certificates whose IP SANs track pod IPs).
This build supports two modes:

* ``cert-manager`` — creates the same Issuer/Certificate CRs for clusters
  that run cert-manager,
* ``self-signed`` (default) — the operator IS the CA: it generates a CA
  keypair plus head/worker leaf certs via openssl and publishes them as the
  Secrets the pod builder mounts (kuberay_amd/common/pod.py configure_tls).
  Leaf certs carry DNS SANs for the head service FQDN + 127.0.0.1, and are
  re-issued when pod IPs appear that are not yet in the SAN set (the
  reference's IP-SAN tracking :279-448).
"""
from __future__ import annotations

import base64
import logging
import os
import subprocess
import tempfile
from typing import Dict, List, Optional, Tuple

from ..common import association
from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..kube.controller import Reconciler, Request, Result
from ..kube.events import EventRecorder, NullRecorder
from ..kube.store import AlreadyExistsError
from ..models import RayCluster
from ..utils import constants as C
from ..utils import names

logger = logging.getLogger("kuberay.mtls")


def _openssl(*args: str, input_: Optional[bytes] = None) -> bytes:
    out = subprocess.run(["openssl", *args], input=input_,
                         capture_output=True, timeout=30)
    if out.returncode != 0:
        raise RuntimeError(f"openssl {' '.join(args[:3])} failed: "
                           f"{out.stderr.decode()[:300]}")
    return out.stdout


def generate_ca(common_name: str = "kuberay-amd-ray-ca") -> Tuple[bytes, bytes]:
    """Returns (ca_cert_pem, ca_key_pem)."""
    with tempfile.TemporaryDirectory() as d:
        key = os.path.join(d, "ca.key")
        crt = os.path.join(d, "ca.crt")
        _openssl("req", "-x509", "-newkey", "rsa:2048", "-nodes",
                 "-keyout", key, "-out", crt, "-days", "3650",
                 "-subj", f"/CN={common_name}")
        return open(crt, "rb").read(), open(key, "rb").read()


def generate_leaf(ca_cert: bytes, ca_key: bytes, common_name: str,
                  dns_sans: List[str], ip_sans: List[str]) -> Tuple[bytes, bytes]:
    """Returns (cert_pem, key_pem) signed by the CA with the given SANs."""
    san = ",".join([f"DNS:{d}" for d in dns_sans] + [f"IP:{i}" for i in ip_sans])
    with tempfile.TemporaryDirectory() as d:
        key = os.path.join(d, "leaf.key")
        csr = os.path.join(d, "leaf.csr")
        crt = os.path.join(d, "leaf.crt")
        ca_crt = os.path.join(d, "ca.crt")
        ca_k = os.path.join(d, "ca.key")
        open(ca_crt, "wb").write(ca_cert)
        open(ca_k, "wb").write(ca_key)
        ext = os.path.join(d, "ext.cnf")
        open(ext, "w").write(f"subjectAltName={san}\n")
        _openssl("req", "-newkey", "rsa:2048", "-nodes", "-keyout", key,
                 "-out", csr, "-subj", f"/CN={common_name}")
        _openssl("x509", "-req", "-in", csr, "-CA", ca_crt, "-CAkey", ca_k,
                 "-CAcreateserial", "-out", crt, "-days", "825",
                 "-extfile", ext)
        return open(crt, "rb").read(), open(key, "rb").read()


def cert_sans(cert_pem: bytes) -> str:
    return _openssl("x509", "-noout", "-ext", "subjectAltName",
                    input_=cert_pem).decode()


class MTLSReconciler(Reconciler):
    """Issues/rotates the ray-tls Secrets for TLS-enabled RayClusters."""

    def __init__(self, client: KubeClient, recorder: Optional[EventRecorder] = None,
                 mode: str = "self-signed"):
        self.client = client
        self.recorder = recorder or NullRecorder()
        self.mode = mode

    def reconcile(self, request: Request) -> Result:
        namespace, name = request
        cluster = self.client.try_get(RayCluster, namespace, name)
        if cluster is None or cluster.metadata.deletion_timestamp:
            return Result()
        if not (cluster.spec.tls_options and cluster.spec.tls_options.enabled):
            return Result()
        if self.mode == "cert-manager":
            self._reconcile_cert_manager(cluster)
            return Result()
        return self._reconcile_self_signed(cluster)

    # ------------------------------------------------------------------
    def _secret(self, namespace: str, name: str) -> Optional[k8s.Secret]:
        return self.client.try_get(k8s.Secret, namespace, name)

    def _create_secret(self, cluster: RayCluster, name: str,
                       data: Dict[str, bytes]) -> None:
        secret = k8s.Secret(
            metadata=k8s.ObjectMeta(
                name=name, namespace=cluster.metadata.namespace or "default",
                labels={C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name},
                owner_references=[k8s.owner_reference_for(cluster)]),
            type="kubernetes.io/tls" if "tls.crt" in data else "Opaque",
            data={k: base64.b64encode(v).decode() for k, v in data.items()},
        )
        try:
            self.client.create(secret)
        except AlreadyExistsError:
            pass

    def _reconcile_self_signed(self, cluster: RayCluster) -> Result:
        namespace = cluster.metadata.namespace or "default"
        cname = cluster.metadata.name
        ca_secret_name = f"ca-secret-{cname}"
        ca_secret = self._secret(namespace, ca_secret_name)
        if ca_secret is None:
            ca_cert, ca_key = generate_ca(f"ray-ca-{cname}")
            self._create_secret(cluster, ca_secret_name,
                                {"ca.crt": ca_cert, "ca.key": ca_key})
            self.recorder.eventf(cluster, "Normal", "GeneratedCA",
                                 "Generated self-signed CA for mTLS")
            ca_secret = self._secret(namespace, ca_secret_name)
            if ca_secret is None:
                return Result(requeue_after=2)
        ca_cert = base64.b64decode(ca_secret.data["ca.crt"])
        ca_key = base64.b64decode(ca_secret.data["ca.key"])

        head_svc = names.head_service_name(C.KIND_RAYCLUSTER, cluster.spec, cname)
        fqdn = names.fqdn_service_name(cluster, namespace)
        dns_sans = [head_svc, fqdn, "localhost"]

        # Per-role pod IP SANs (reference reconcileHeadCertificate /
        # reconcileWorkerCertificate, raycluster_mtls_controller.go:279-448):
        # GCS dials WORKERS by pod IP, so worker certs must carry worker pod
        # IPs; the head cert carries only head pod IPs so a worker scale
        # event never forces a head cert reissue.
        head_ips: List[str] = ["127.0.0.1"]
        worker_ips: List[str] = ["127.0.0.1"]
        views = getattr(self.client, "list_pod_views", None)
        if views is not None:
            for v in views(namespace,
                           association.cluster_head_pod_selector(cname)):
                if v.pod_ip:
                    head_ips.append(v.pod_ip)
            for v in views(namespace,
                           association.cluster_worker_pods_selector(cname)):
                if v.pod_ip:
                    worker_ips.append(v.pod_ip)

        changed = False
        for role, secret_name, want_ips in (
                ("head", f"ray-head-secret-{cname}", sorted(set(head_ips))),
                ("worker", f"ray-worker-secret-{cname}", sorted(set(worker_ips)))):
            existing = self._secret(namespace, secret_name)
            if existing is not None:
                covered = self._sans_cover(existing, want_ips)
                if covered:
                    continue
                self.client.delete(existing)
            cert, key = generate_leaf(ca_cert, ca_key, f"ray-{role}-{cname}",
                                      dns_sans, want_ips)
            self._create_secret(cluster, secret_name, {
                "tls.crt": cert, "tls.key": key, "ca.crt": ca_cert})
            changed = True
        if changed:
            self.recorder.eventf(cluster, "Normal", "IssuedTLSCertificates",
                                 "Issued mTLS leaf certificates")
        return Result(requeue_after=30)

    @staticmethod
    def _sans_cover(secret: k8s.Secret, want_ips: List[str]) -> bool:
        """Exact-entry SAN membership: substring matching would let
        10.0.0.11 falsely 'cover' 10.0.0.1 and skip a needed reissue."""
        try:
            cert = base64.b64decode(secret.data["tls.crt"])
            sans_text = cert_sans(cert)
        except Exception:
            return False
        entries = set()
        for part in sans_text.replace("\n", ",").split(","):
            part = part.strip()
            if part.startswith("IP Address:"):
                entries.add(part[len("IP Address:"):].strip())
            elif part.startswith("IP:"):
                entries.add(part[len("IP:"):].strip())
        return all(ip in entries for ip in want_ips)

    # ------------------------------------------------------------------
    def _reconcile_cert_manager(self, cluster: RayCluster) -> None:
        """Create cert-manager CRs (Issuer → CA cert → CA Issuer → leaves)."""
        namespace = cluster.metadata.namespace or "default"
        cname = cluster.metadata.name
        from ..kube.client import RawObjectClient
        raw = RawObjectClient(self.client)
        owner = [k8s.owner_reference_for(cluster).to_dict()]
        # Per-role IP SANs, mirroring reconcileHeadCertificate /
        # reconcileWorkerCertificate: worker cert tracks worker pod IPs
        # (GCS dials workers by IP), head cert tracks head pod IPs only.
        head_ips: List[str] = ["127.0.0.1"]
        worker_ips: List[str] = ["127.0.0.1"]
        views = getattr(self.client, "list_pod_views", None)
        if views is not None:
            for v in views(namespace,
                           association.cluster_head_pod_selector(cname)):
                if v.pod_ip:
                    head_ips.append(v.pod_ip)
            for v in views(namespace,
                           association.cluster_worker_pods_selector(cname)):
                if v.pod_ip:
                    worker_ips.append(v.pod_ip)
        objs = [
            {"apiVersion": "cert-manager.io/v1", "kind": "Issuer",
             "metadata": {"name": f"ray-selfsigned-issuer-{cname}",
                          "namespace": namespace, "ownerReferences": owner},
             "spec": {"selfSigned": {}}},
            {"apiVersion": "cert-manager.io/v1", "kind": "Certificate",
             "metadata": {"name": f"ray-ca-certificate-{cname}",
                          "namespace": namespace, "ownerReferences": owner},
             "spec": {"isCA": True, "commonName": f"ray-ca-{cname}",
                      "secretName": f"ca-secret-{cname}",
                      "issuerRef": {"name": f"ray-selfsigned-issuer-{cname}",
                                    "kind": "Issuer"}}},
            {"apiVersion": "cert-manager.io/v1", "kind": "Issuer",
             "metadata": {"name": f"ray-ca-issuer-{cname}",
                          "namespace": namespace, "ownerReferences": owner},
             "spec": {"ca": {"secretName": f"ca-secret-{cname}"}}},
            {"apiVersion": "cert-manager.io/v1", "kind": "Certificate",
             "metadata": {"name": f"ray-head-cert-{cname}",
                          "namespace": namespace, "ownerReferences": owner},
             "spec": {"secretName": f"ray-head-secret-{cname}",
                      "dnsNames": [names.fqdn_service_name(cluster, namespace)],
                      "ipAddresses": sorted(set(head_ips)),
                      "issuerRef": {"name": f"ray-ca-issuer-{cname}",
                                    "kind": "Issuer"}}},
            {"apiVersion": "cert-manager.io/v1", "kind": "Certificate",
             "metadata": {"name": f"ray-worker-cert-{cname}",
                          "namespace": namespace, "ownerReferences": owner},
             "spec": {"secretName": f"ray-worker-secret-{cname}",
                      "ipAddresses": sorted(set(worker_ips)),
                      "issuerRef": {"name": f"ray-ca-issuer-{cname}",
                                    "kind": "Issuer"}}},
        ]
        for obj in objs:
            try:
                raw.create(obj)
            except AlreadyExistsError:
                # Certificates: keep ipAddresses tracking the live pod IPs
                # (the reference updates the Certificate spec in place).
                if obj["kind"] == "Certificate" and "ipAddresses" in obj["spec"]:
                    md = obj["metadata"]
                    cur = raw.try_get("Certificate", md["namespace"],
                                      md["name"],
                                      api_version="cert-manager.io/v1")
                    if cur is not None and (cur.get("spec", {}).get("ipAddresses")
                                            != obj["spec"]["ipAddresses"]):
                        cur["spec"]["ipAddresses"] = obj["spec"]["ipAddresses"]
                        raw.update(cur)
