"""Service construction (reference: ray-operator/controllers/ray/common/service.go).

* head Service — ports taken from the Ray container's named ports, falling
  back to the defaults (gcs 6379, dashboard 8265, client 10001, metrics 8080,
  serve 8000; service.go:436-447 getDefaultPorts),
* serve Service — selector targets healthy serve proxies via the
  ``ray.io/serve=true`` label (service.go:181),
* headless Service — for multi-host worker groups (service.go:299).
"""
from __future__ import annotations

import os
from typing import Dict, Optional

from ..kube.objects import Service, ServicePort, ServiceSpec, ObjectMeta
from ..models.raycluster import RayCluster, RayNodeType
from ..utils import constants as C
from ..utils import names

DEFAULT_PORTS = {
    C.GCS_SERVER_PORT_NAME: C.DEFAULT_GCS_SERVER_PORT,
    C.DASHBOARD_PORT_NAME: C.DEFAULT_DASHBOARD_PORT,
    C.CLIENT_PORT_NAME: C.DEFAULT_CLIENT_PORT,
    C.METRICS_PORT_NAME: C.DEFAULT_METRICS_PORT,
    C.SERVING_PORT_NAME: C.DEFAULT_SERVING_PORT,
}


def _ports_from_cluster(cluster: RayCluster) -> Dict[str, int]:
    ports = dict(DEFAULT_PORTS)
    container = cluster.spec.head_group_spec.template.spec.containers[C.RAY_CONTAINER_INDEX]
    for p in container.ports or []:
        if p.name and p.container_port:
            ports[p.name] = p.container_port
    # rayStartParams port overrides the gcs port
    params = cluster.spec.head_group_spec.ray_start_params or {}
    if "port" in params:
        try:
            ports[C.GCS_SERVER_PORT_NAME] = int(params["port"])
        except ValueError:
            pass
    return ports


def build_head_service(cluster: RayCluster, creator_crd_type: Optional[str] = None,
                       owner_name: Optional[str] = None) -> Service:
    """service.go:37 BuildServiceForHeadPod."""
    owner_name = owner_name or cluster.metadata.name
    crd_type = creator_crd_type or C.KIND_RAYCLUSTER
    name = names.head_service_name(crd_type, cluster.spec, owner_name)
    namespace = cluster.metadata.namespace or "default"

    selector = {
        C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name,
        C.RAY_NODE_TYPE_LABEL_KEY: RayNodeType.HEAD,
        C.RAY_ID_LABEL_KEY: names.check_label(
            names.identifier(cluster.metadata.name, RayNodeType.HEAD)),
    }
    labels = {
        C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name,
        C.RAY_NODE_TYPE_LABEL_KEY: RayNodeType.HEAD,
        C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
        C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
    }

    ports = [
        ServicePort(name=pname, port=pnum, target_port=pnum,
                    app_protocol=C.DEFAULT_SERVICE_APP_PROTOCOL)
        for pname, pnum in sorted(_ports_from_cluster(cluster).items())
    ]

    # Headless by default so GCS clients talk pod-direct (xGMI-local traffic
    # is node-internal anyway on the single 8xMI355X node); a normal
    # ClusterIP service via env flag (reference: ENABLE_RAY_HEAD_CLUSTER_IP_SERVICE).
    cluster_ip: Optional[str] = "None"
    if os.environ.get(C.ENABLE_RAY_HEAD_CLUSTER_IP_SERVICE, "").lower() == "true":
        cluster_ip = None

    svc = Service(
        metadata=ObjectMeta(name=name, namespace=namespace, labels=labels,
                            annotations=dict(cluster.spec.head_service_annotations or {})),
        spec=ServiceSpec(selector=selector, ports=ports, cluster_ip=cluster_ip,
                         type=cluster.spec.head_group_spec.service_type),
    )
    # user-provided headService template merges name/labels/annotations
    user_svc = cluster.spec.head_group_spec.head_service
    if user_svc is not None:
        if user_svc.metadata.labels:
            svc.metadata.labels.update(user_svc.metadata.labels)
        if user_svc.metadata.annotations:
            svc.metadata.ensure_annotations().update(user_svc.metadata.annotations)
        if user_svc.spec.type:
            svc.spec.type = user_svc.spec.type
    return svc


def build_serve_service(owner, cluster: RayCluster,
                        is_rayservice: bool = True) -> Service:
    """service.go:181 BuildServeService — selector includes the serve-traffic
    readiness label managed by the RayService controller."""
    owner_name = owner.metadata.name
    name = names.serve_service_name(owner_name)
    namespace = cluster.metadata.namespace or "default"
    serve_port = DEFAULT_PORTS[C.SERVING_PORT_NAME]
    container = cluster.spec.head_group_spec.template.spec.containers[C.RAY_CONTAINER_INDEX]
    for p in container.ports or []:
        if p.name == C.SERVING_PORT_NAME and p.container_port:
            serve_port = p.container_port

    selector = {
        C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name,
        C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY: C.ENABLE_RAY_CLUSTER_SERVING_SERVICE_TRUE,
    }
    labels = {
        C.RAY_ORIGINATED_FROM_CR_NAME_LABEL_KEY: names.check_label(owner_name),
        C.RAY_ORIGINATED_FROM_CRD_LABEL_KEY: owner.kind,
        C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
        C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
    }
    svc = Service(
        metadata=ObjectMeta(name=name, namespace=namespace, labels=labels),
        spec=ServiceSpec(selector=selector, ports=[
            ServicePort(name=C.SERVING_PORT_NAME, port=serve_port,
                        target_port=serve_port,
                        app_protocol=C.DEFAULT_SERVICE_APP_PROTOCOL)]),
    )
    user_svc = getattr(owner.spec, "serve_service", None)
    if user_svc is not None:
        if user_svc.metadata.name:
            svc.metadata.name = user_svc.metadata.name
        if user_svc.metadata.labels:
            svc.metadata.labels.update(user_svc.metadata.labels)
        if user_svc.spec.ports:
            svc.spec.ports = user_svc.spec.ports
        if user_svc.spec.type:
            svc.spec.type = user_svc.spec.type
    return svc


def build_headless_service(cluster: RayCluster) -> Service:
    """service.go:299 BuildHeadlessServiceForRayCluster — multi-host worker
    group pod-to-pod DNS."""
    name = names.headless_service_name(cluster.metadata.name)
    namespace = cluster.metadata.namespace or "default"
    return Service(
        metadata=ObjectMeta(
            name=name, namespace=namespace,
            labels={
                C.RAY_CLUSTER_HEADLESS_SERVICE_LABEL_KEY: cluster.metadata.name,
                C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
                C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
            }),
        spec=ServiceSpec(
            cluster_ip="None",
            publish_not_ready_addresses=True,
            selector={
                C.RAY_CLUSTER_LABEL_KEY: cluster.metadata.name,
                C.RAY_NODE_TYPE_LABEL_KEY: RayNodeType.WORKER,
            },
            ports=[],
        ),
    )
