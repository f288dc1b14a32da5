"""Ingress construction for the head dashboard (reference: common/ingress.go)."""
from __future__ import annotations

from ..kube.objects import Ingress, ObjectMeta
from ..models.raycluster import RayCluster
from ..utils import constants as C
from ..utils import names


def build_ingress_for_head_service(cluster: RayCluster) -> Ingress:
    """ingress.go BuildIngressForHeadService — routes / to the dashboard port."""
    head_svc = names.head_service_name(C.KIND_RAYCLUSTER, cluster.spec, cluster.metadata.name)
    opts = cluster.spec.head_group_spec.ingress_options or {}
    host = opts.get("host")
    path = opts.get("path", "/" + cluster.metadata.name + "/(.*)")
    path_type = opts.get("pathType", "Exact" if opts.get("path") else "ImplementationSpecific")
    annotations = {}
    if not opts.get("path"):
        annotations["nginx.ingress.kubernetes.io/rewrite-target"] = "/$1"
    rule = {
        "http": {
            "paths": [{
                "path": path,
                "pathType": path_type,
                "backend": {
                    "service": {
                        "name": head_svc,
                        "port": {"number": C.DEFAULT_DASHBOARD_PORT},
                    }
                },
            }]
        }
    }
    if host:
        rule["host"] = host
    spec = {"rules": [rule]}
    if opts.get("ingressClassName"):
        spec["ingressClassName"] = opts["ingressClassName"]
    if opts.get("tls"):
        spec["tls"] = opts["tls"]
    return Ingress(
        metadata=ObjectMeta(
            name=names.ingress_name(cluster.metadata.name),
            namespace=cluster.metadata.namespace or "default",
            labels={
                C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name,
                C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
                C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
            },
            annotations=annotations or None,
        ),
        spec=spec,
    )
