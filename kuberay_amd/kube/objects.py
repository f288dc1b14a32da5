"""Typed Kubernetes object model (the client-go analog).

Pydantic v2 models with camelCase wire names and ``extra="allow"`` so that
arbitrary user-supplied pod templates round-trip losslessly even where we
don't model a field. Only the kinds and fields the operator actually
manipulates are fully typed; everything else passes through untouched.

Reference analog: k8s.io/api/core/v1 + batch/v1 as consumed by
ray-operator/controllers/ray/common/*.go.
"""
from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional

from pydantic import BaseModel, ConfigDict, Field
from pydantic.alias_generators import to_camel


class K8sModel(BaseModel):
    """Base for every Kubernetes-shaped object."""

    model_config = ConfigDict(
        alias_generator=to_camel,
        populate_by_name=True,
        extra="allow",
        validate_assignment=False,
    )

    def to_dict(self) -> Dict[str, Any]:
        return self.model_dump(by_alias=True, exclude_none=True)

    @classmethod
    def from_dict(cls, data: Dict[str, Any]):
        return cls.model_validate(data)

    def clone(self):
        return copy.deepcopy(self)


# ---------------------------------------------------------------------------
# metav1
# ---------------------------------------------------------------------------

class OwnerReference(K8sModel):
    api_version: Optional[str] = None
    kind: Optional[str] = None
    name: Optional[str] = None
    uid: Optional[str] = None
    controller: Optional[bool] = None
    block_owner_deletion: Optional[bool] = None


class ObjectMeta(K8sModel):
    name: Optional[str] = None
    generate_name: Optional[str] = None
    namespace: Optional[str] = None
    labels: Optional[Dict[str, str]] = None
    annotations: Optional[Dict[str, str]] = None
    uid: Optional[str] = None
    resource_version: Optional[str] = None
    generation: Optional[int] = None
    creation_timestamp: Optional[str] = None
    deletion_timestamp: Optional[str] = None
    finalizers: Optional[List[str]] = None
    owner_references: Optional[List[OwnerReference]] = None

    def ensure_labels(self) -> Dict[str, str]:
        if self.labels is None:
            self.labels = {}
        return self.labels

    def ensure_annotations(self) -> Dict[str, str]:
        if self.annotations is None:
            self.annotations = {}
        return self.annotations


class Condition(K8sModel):
    """metav1.Condition."""

    type: Optional[str] = None
    status: Optional[str] = None  # "True" | "False" | "Unknown"
    reason: Optional[str] = None
    message: Optional[str] = None
    last_transition_time: Optional[str] = None
    observed_generation: Optional[int] = None


class LabelSelector(K8sModel):
    match_labels: Optional[Dict[str, str]] = None
    match_expressions: Optional[List[Dict[str, Any]]] = None


# ---------------------------------------------------------------------------
# corev1 — containers & pods
# ---------------------------------------------------------------------------

class EnvVarSource(K8sModel):
    field_ref: Optional[Dict[str, Any]] = None
    secret_key_ref: Optional[Dict[str, Any]] = None
    config_map_key_ref: Optional[Dict[str, Any]] = None
    resource_field_ref: Optional[Dict[str, Any]] = None


class EnvVar(K8sModel):
    name: str
    value: Optional[str] = None
    value_from: Optional[EnvVarSource] = None


class ContainerPort(K8sModel):
    name: Optional[str] = None
    container_port: Optional[int] = None
    protocol: Optional[str] = None
    host_port: Optional[int] = None


class VolumeMount(K8sModel):
    name: str
    mount_path: Optional[str] = None
    read_only: Optional[bool] = None
    sub_path: Optional[str] = None


class ResourceRequirements(K8sModel):
    limits: Optional[Dict[str, Any]] = None
    requests: Optional[Dict[str, Any]] = None


class ExecAction(K8sModel):
    command: Optional[List[str]] = None


class HTTPGetAction(K8sModel):
    path: Optional[str] = None
    port: Optional[Any] = None
    host: Optional[str] = None
    scheme: Optional[str] = None


class TCPSocketAction(K8sModel):
    port: Optional[Any] = None


class Probe(K8sModel):
    exec_: Optional[ExecAction] = Field(default=None, alias="exec")
    http_get: Optional[HTTPGetAction] = None
    tcp_socket: Optional[TCPSocketAction] = None
    initial_delay_seconds: Optional[int] = None
    timeout_seconds: Optional[int] = None
    period_seconds: Optional[int] = None
    success_threshold: Optional[int] = None
    failure_threshold: Optional[int] = None


class SecurityContext(K8sModel):
    privileged: Optional[bool] = None
    capabilities: Optional[Dict[str, Any]] = None
    run_as_user: Optional[int] = None
    run_as_group: Optional[int] = None


class Lifecycle(K8sModel):
    post_start: Optional[Dict[str, Any]] = None
    pre_stop: Optional[Dict[str, Any]] = None


class Container(K8sModel):
    name: str
    image: Optional[str] = None
    command: Optional[List[str]] = None
    args: Optional[List[str]] = None
    working_dir: Optional[str] = None
    env: Optional[List[EnvVar]] = None
    ports: Optional[List[ContainerPort]] = None
    resources: Optional[ResourceRequirements] = None
    volume_mounts: Optional[List[VolumeMount]] = None
    liveness_probe: Optional[Probe] = None
    readiness_probe: Optional[Probe] = None
    startup_probe: Optional[Probe] = None
    security_context: Optional[SecurityContext] = None
    lifecycle: Optional[Lifecycle] = None
    image_pull_policy: Optional[str] = None

    # -- env helpers ---------------------------------------------------
    def env_names(self) -> List[str]:
        return [e.name for e in (self.env or [])]

    def get_env(self, name: str) -> Optional[EnvVar]:
        for e in self.env or []:
            if e.name == name:
                return e
        return None

    def set_env_if_absent(self, name: str, value: str) -> None:
        if self.env is None:
            self.env = []
        if self.get_env(name) is None:
            self.env.append(EnvVar(name=name, value=value))

    def set_env(self, name: str, value: str) -> None:
        if self.env is None:
            self.env = []
        existing = self.get_env(name)
        if existing is not None:
            existing.value = value
            existing.value_from = None
        else:
            self.env.append(EnvVar(name=name, value=value))

    def add_volume_mount_if_absent(self, mount: VolumeMount) -> None:
        if self.volume_mounts is None:
            self.volume_mounts = []
        for m in self.volume_mounts:
            if m.name == mount.name:
                return
        self.volume_mounts.append(mount)


class Volume(K8sModel):
    name: str
    empty_dir: Optional[Dict[str, Any]] = None
    host_path: Optional[Dict[str, Any]] = None
    config_map: Optional[Dict[str, Any]] = None
    secret: Optional[Dict[str, Any]] = None
    persistent_volume_claim: Optional[Dict[str, Any]] = None
    projected: Optional[Dict[str, Any]] = None


class Toleration(K8sModel):
    key: Optional[str] = None
    operator: Optional[str] = None
    value: Optional[str] = None
    effect: Optional[str] = None
    toleration_seconds: Optional[int] = None


class PodSpec(K8sModel):
    containers: List[Container] = Field(default_factory=list)
    init_containers: Optional[List[Container]] = None
    volumes: Optional[List[Volume]] = None
    node_selector: Optional[Dict[str, str]] = None
    affinity: Optional[Dict[str, Any]] = None
    tolerations: Optional[List[Toleration]] = None
    service_account_name: Optional[str] = None
    restart_policy: Optional[str] = None
    priority_class_name: Optional[str] = None
    scheduler_name: Optional[str] = None
    termination_grace_period_seconds: Optional[int] = None
    host_network: Optional[bool] = None
    dns_policy: Optional[str] = None
    image_pull_secrets: Optional[List[Dict[str, Any]]] = None
    security_context: Optional[Dict[str, Any]] = None

    def add_volume_if_absent(self, volume: Volume) -> None:
        if self.volumes is None:
            self.volumes = []
        for v in self.volumes:
            if v.name == volume.name:
                return
        self.volumes.append(volume)


class PodTemplateSpec(K8sModel):
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: PodSpec = Field(default_factory=PodSpec)


class ContainerStateTerminated(K8sModel):
    exit_code: Optional[int] = None
    reason: Optional[str] = None
    finished_at: Optional[str] = None


class ContainerState(K8sModel):
    waiting: Optional[Dict[str, Any]] = None
    running: Optional[Dict[str, Any]] = None
    terminated: Optional[ContainerStateTerminated] = None


class ContainerStatus(K8sModel):
    name: Optional[str] = None
    ready: Optional[bool] = None
    restart_count: Optional[int] = None
    state: Optional[ContainerState] = None
    last_state: Optional[ContainerState] = None


class PodCondition(K8sModel):
    type: Optional[str] = None
    status: Optional[str] = None
    reason: Optional[str] = None
    message: Optional[str] = None
    last_transition_time: Optional[str] = None


class PodStatus(K8sModel):
    phase: Optional[str] = None  # Pending/Running/Succeeded/Failed/Unknown
    conditions: Optional[List[PodCondition]] = None
    pod_ip: Optional[str] = Field(default=None, alias="podIP")
    host_ip: Optional[str] = Field(default=None, alias="hostIP")
    container_statuses: Optional[List[ContainerStatus]] = None
    init_container_statuses: Optional[List[ContainerStatus]] = None
    reason: Optional[str] = None
    message: Optional[str] = None
    start_time: Optional[str] = None


class Pod(K8sModel):
    api_version: str = "v1"
    kind: str = "Pod"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: PodSpec = Field(default_factory=PodSpec)
    status: PodStatus = Field(default_factory=PodStatus)


# ---------------------------------------------------------------------------
# corev1 — services & friends
# ---------------------------------------------------------------------------

class ServicePort(K8sModel):
    name: Optional[str] = None
    port: Optional[int] = None
    target_port: Optional[Any] = None
    protocol: Optional[str] = None
    app_protocol: Optional[str] = None
    node_port: Optional[int] = None


class ServiceSpec(K8sModel):
    selector: Optional[Dict[str, str]] = None
    ports: Optional[List[ServicePort]] = None
    type: Optional[str] = None
    cluster_ip: Optional[str] = Field(default=None, alias="clusterIP")
    publish_not_ready_addresses: Optional[bool] = None


class Service(K8sModel):
    api_version: str = "v1"
    kind: str = "Service"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: ServiceSpec = Field(default_factory=ServiceSpec)
    status: Optional[Dict[str, Any]] = None


class Secret(K8sModel):
    api_version: str = "v1"
    kind: str = "Secret"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    type: Optional[str] = None
    data: Optional[Dict[str, str]] = None
    string_data: Optional[Dict[str, str]] = None


class ConfigMap(K8sModel):
    api_version: str = "v1"
    kind: str = "ConfigMap"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    data: Optional[Dict[str, str]] = None


class PersistentVolumeClaim(K8sModel):
    api_version: str = "v1"
    kind: str = "PersistentVolumeClaim"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: Optional[Dict[str, Any]] = None
    status: Optional[Dict[str, Any]] = None


class ServiceAccount(K8sModel):
    api_version: str = "v1"
    kind: str = "ServiceAccount"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)


class PolicyRule(K8sModel):
    api_groups: Optional[List[str]] = None
    resources: Optional[List[str]] = None
    verbs: Optional[List[str]] = None
    resource_names: Optional[List[str]] = None


class Role(K8sModel):
    api_version: str = "rbac.authorization.k8s.io/v1"
    kind: str = "Role"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    rules: Optional[List[PolicyRule]] = None


class RoleBinding(K8sModel):
    api_version: str = "rbac.authorization.k8s.io/v1"
    kind: str = "RoleBinding"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    subjects: Optional[List[Dict[str, Any]]] = None
    role_ref: Optional[Dict[str, Any]] = None


class Endpoint(K8sModel):
    addresses: Optional[List[str]] = None
    conditions: Optional[Dict[str, Any]] = None
    target_ref: Optional[Dict[str, Any]] = None


class EndpointSlice(K8sModel):
    api_version: str = "discovery.k8s.io/v1"
    kind: str = "EndpointSlice"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    address_type: Optional[str] = None
    endpoints: Optional[List[Endpoint]] = None
    ports: Optional[List[Dict[str, Any]]] = None


class NetworkPolicy(K8sModel):
    api_version: str = "networking.k8s.io/v1"
    kind: str = "NetworkPolicy"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: Optional[Dict[str, Any]] = None


class Ingress(K8sModel):
    api_version: str = "networking.k8s.io/v1"
    kind: str = "Ingress"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: Optional[Dict[str, Any]] = None
    status: Optional[Dict[str, Any]] = None


class Event(K8sModel):
    api_version: str = "v1"
    kind: str = "Event"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    type: Optional[str] = None  # Normal | Warning
    reason: Optional[str] = None
    message: Optional[str] = None
    involved_object: Optional[Dict[str, Any]] = None
    count: Optional[int] = None
    first_timestamp: Optional[str] = None
    last_timestamp: Optional[str] = None


# ---------------------------------------------------------------------------
# batchv1
# ---------------------------------------------------------------------------

class JobSpec(K8sModel):
    template: PodTemplateSpec = Field(default_factory=PodTemplateSpec)
    backoff_limit: Optional[int] = None
    active_deadline_seconds: Optional[int] = None
    ttl_seconds_after_finished: Optional[int] = None
    completions: Optional[int] = None
    parallelism: Optional[int] = None


class JobCondition(K8sModel):
    type: Optional[str] = None  # Complete | Failed
    status: Optional[str] = None
    reason: Optional[str] = None
    message: Optional[str] = None


class JobStatus(K8sModel):
    active: Optional[int] = None
    succeeded: Optional[int] = None
    failed: Optional[int] = None
    conditions: Optional[List[JobCondition]] = None
    start_time: Optional[str] = None
    completion_time: Optional[str] = None


class Job(K8sModel):
    api_version: str = "batch/v1"
    kind: str = "Job"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: JobSpec = Field(default_factory=JobSpec)
    status: Optional[JobStatus] = None


# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------

def owner_reference_for(obj: Any, *, controller: bool = True) -> OwnerReference:
    """Build a controller OwnerReference for a typed or dict-shaped object."""
    if isinstance(obj, dict):
        meta = obj.get("metadata", {})
        return OwnerReference(
            api_version=obj.get("apiVersion"),
            kind=obj.get("kind"),
            name=meta.get("name"),
            uid=meta.get("uid"),
            controller=controller,
            block_owner_deletion=True,
        )
    return OwnerReference(
        api_version=getattr(obj, "api_version", None),
        kind=getattr(obj, "kind", None),
        name=obj.metadata.name,
        uid=obj.metadata.uid,
        controller=controller,
        block_owner_deletion=True,
    )


def is_owned_by(obj: Any, owner_uid: str) -> bool:
    meta = obj.metadata if not isinstance(obj, dict) else ObjectMeta.from_dict(obj.get("metadata", {}))
    for ref in meta.owner_references or []:
        if ref.uid == owner_uid:
            return True
    return False
