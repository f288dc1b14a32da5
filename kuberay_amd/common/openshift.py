"""OpenShift Route construction (reference: common/openshift.go:1-66).

On OpenShift the head dashboard is exposed as a Route instead of an
Ingress; selection happens in the RayCluster reconciler based on platform
detection (reference main.go:275-306) or the USE_INGRESS_ON_OPENSHIFT env.
"""
from __future__ import annotations

import os

from typing import Optional

from ..kube.objects import K8sModel, ObjectMeta
from ..models.raycluster import RayCluster
from ..utils import constants as C
from ..utils import names


class Route(K8sModel):
    api_version: str = "route.openshift.io/v1"
    kind: str = "Route"
    metadata: Optional[ObjectMeta] = None
    spec: Optional[dict] = None


def use_ingress_on_openshift() -> bool:
    return os.environ.get("USE_INGRESS_ON_OPENSHIFT", "").lower() == "true"


def build_route_for_head_service(cluster: RayCluster) -> Route:
    """openshift.go BuildRouteForHeadService — dashboard route."""
    head_svc = names.head_service_name(C.KIND_RAYCLUSTER, cluster.spec,
                                       cluster.metadata.name)
    return Route(
        metadata=ObjectMeta(
            name=f"{cluster.metadata.name}-{C.HEAD_NODE}-route",
            namespace=cluster.metadata.namespace or "default",
            labels={
                C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name,
                C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
                C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
            }),
        spec={
            "to": {"kind": "Service", "name": head_svc},
            "port": {"targetPort": C.DEFAULT_DASHBOARD_PORT},
        },
    )
