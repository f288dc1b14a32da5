"""Replica & resource calculators (reference: utils/util.go:389-570,
utils/resources.go:8-17).

MI355X scoping: GPU detection is keyed on ``amd.com/gpu`` ONLY. The
reference's generic ``*gpu`` suffix matching plus the NVIDIA MIG regex
(resources.go:8-17) are intentionally absent — no dual-vendor dispatch.
"""
from __future__ import annotations

from decimal import Decimal
from typing import Dict, List

from ..kube.objects import Pod
from ..models.raycluster import RayCluster, WorkerGroupSpec
from . import constants as C
from .quantity import format_quantity, parse_quantity

INT32_MAX = 2147483647


def is_amd_gpu_resource(key: str) -> bool:
    """The only GPU resource key this operator understands."""
    return key == C.AMD_GPU_RESOURCE_NAME


def worker_group_desired_replicas(group: WorkerGroupSpec) -> int:
    """util.go:389 GetWorkerGroupDesiredReplicas — clamp replicas to [min,max],
    suspended groups count 0."""
    if group.suspend:
        return 0
    min_r = group.min_replicas if group.min_replicas is not None else 0
    max_r = group.max_replicas if group.max_replicas is not None else INT32_MAX
    replicas = group.replicas
    if replicas is None or replicas < min_r:
        replicas = min_r
    elif replicas > max_r:
        replicas = max_r
    return replicas


def calculate_desired_replicas(cluster: RayCluster) -> int:
    return sum(
        worker_group_desired_replicas(g) * max(g.num_of_hosts, 1)
        for g in cluster.spec.worker_group_specs
    )


def calculate_min_replicas(cluster: RayCluster) -> int:
    total = 0
    for g in cluster.spec.worker_group_specs:
        if g.suspend:
            continue
        total += (g.min_replicas or 0) * max(g.num_of_hosts, 1)
    return total


def calculate_max_replicas(cluster: RayCluster) -> int:
    total = 0
    for g in cluster.spec.worker_group_specs:
        if g.suspend:
            continue
        max_r = g.max_replicas if g.max_replicas is not None else INT32_MAX
        total += max_r * max(g.num_of_hosts, 1)
    return min(total, INT32_MAX)


def calculate_available_replicas(pods: List[Pod]) -> int:
    """Pods labeled as workers in Running phase (util.go analog)."""
    count = 0
    for pod in pods:
        labels = pod.metadata.labels or {}
        if labels.get(C.RAY_NODE_TYPE_LABEL_KEY) != C.WORKER_NODE:
            continue
        if pod.status.phase == "Running":
            count += 1
    return count


def calculate_ready_replicas(pods: List[Pod]) -> int:
    count = 0
    for pod in pods:
        labels = pod.metadata.labels or {}
        if labels.get(C.RAY_NODE_TYPE_LABEL_KEY) != C.WORKER_NODE:
            continue
        if is_pod_ready(pod):
            count += 1
    return count


def is_pod_ready(pod: Pod) -> bool:
    for cond in pod.status.conditions or []:
        if cond.type == "Ready" and cond.status == "True":
            return True
    return False


def is_pod_running_and_ready(pod: Pod) -> bool:
    return pod.status.phase == "Running" and is_pod_ready(pod)


def _container_resource(container, key: str) -> Decimal:
    res = container.resources
    if res is None:
        return Decimal(0)
    limits = res.limits or {}
    requests = res.requests or {}
    # K8s semantics: request defaults to limit when only limit set; the
    # reference sums limits (falling back to requests) for desired totals.
    if key in limits:
        return parse_quantity(limits[key])
    if key in requests:
        return parse_quantity(requests[key])
    return Decimal(0)


def _pod_template_resource(template, key: str) -> Decimal:
    return sum(
        (_container_resource(c, key) for c in template.spec.containers),
        Decimal(0),
    )


def calculate_desired_resources(cluster: RayCluster) -> Dict[str, str]:
    """Sum head + workers' cpu/memory/amd.com/gpu into canonical quantities
    (raycluster_controller.go:2409-2420 analog, amd.com/gpu only)."""
    totals = {"cpu": Decimal(0), "memory": Decimal(0), C.AMD_GPU_RESOURCE_NAME: Decimal(0)}
    head_template = cluster.spec.head_group_spec.template
    for key in totals:
        totals[key] += _pod_template_resource(head_template, key)
    for group in cluster.spec.worker_group_specs:
        replicas = worker_group_desired_replicas(group) * max(group.num_of_hosts, 1)
        for key in totals:
            totals[key] += _pod_template_resource(group.template, key) * replicas
    return {
        "desiredCPU": format_quantity(totals["cpu"]),
        "desiredMemory": format_quantity(totals["memory"]),
        "desiredGPU": format_quantity(totals[C.AMD_GPU_RESOURCE_NAME]),
    }


def container_gpu_count(container) -> int:
    """Number of MI355X GPUs a container requests (amd.com/gpu limit)."""
    return int(_container_resource(container, C.AMD_GPU_RESOURCE_NAME))


def find_container_port(container, port_name: str, default_port: int) -> int:
    """util.go:676 FindContainerPort."""
    for port in container.ports or []:
        if port.name == port_name and port.container_port:
            return port.container_port
    return default_port


def pod_gpu_count(pod_or_template) -> int:
    spec = pod_or_template.spec
    return sum(container_gpu_count(c) for c in spec.containers)
