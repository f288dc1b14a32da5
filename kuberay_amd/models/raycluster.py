"""ray.io/v1 RayCluster types.

Field surface mirrors the reference CRD
(ray-operator/apis/ray/v1/raycluster_types.go) so sample YAMLs and clients
work verbatim. Differences from the reference are deliberate MI355X scoping
(SURVEY.md §2.1): there is no ``desiredTPU`` status column and no NVIDIA/TPU
accelerator plumbing anywhere.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from pydantic import Field

from ..kube.objects import (
    Condition,
    EnvVar,
    K8sModel,
    ObjectMeta,
    PodTemplateSpec,
    ResourceRequirements,
    SecurityContext,
    Service,
    VolumeMount,
)
from ..utils import constants as C

# ---------------------------------------------------------------------------
# enums (string constants; raycluster_types.go:556-566, :653-665)
# ---------------------------------------------------------------------------

class ClusterState:
    READY = "ready"
    FAILED = "failed"  # deprecated in the reference but kept for surface parity
    SUSPENDED = "suspended"


class RayClusterConditionType:
    PROVISIONED = "RayClusterProvisioned"
    HEAD_POD_READY = "HeadPodReady"
    REPLICA_FAILURE = "ReplicaFailure"
    SUSPENDING = "RayClusterSuspending"
    SUSPENDED = "RayClusterSuspended"


class RayClusterConditionReason:
    ALL_POD_RUNNING_AND_READY_FIRST_TIME = "AllPodRunningAndReadyFirstTime"
    RAY_CLUSTER_PODS_PROVISIONING = "RayClusterPodsProvisioning"
    HEAD_POD_NOT_FOUND = "HeadPodNotFound"
    HEAD_POD_RUNNING_AND_READY = "HeadPodRunningAndReady"
    UNKNOWN = "Unknown"


class RayNodeType:
    HEAD = "head"
    WORKER = "worker"
    REDIS_CLEANUP = "redis-cleanup"


class UpscalingMode:
    DEFAULT = "Default"
    AGGRESSIVE = "Aggressive"
    CONSERVATIVE = "Conservative"


class AutoscalerVersion:
    V1 = "v1"
    V2 = "v2"


class GcsFaultToleranceBackend:
    REDIS = "redis"
    EMBEDDED = "embedded"


class AuthMode:
    TOKEN = "token"
    DISABLED = "disabled"


class RayClusterUpgradeType:
    RECREATE = "Recreate"
    NONE = "None"


class NetworkPolicyMode:
    DENY_ALL = "DenyAll"
    DENY_ALL_INGRESS = "DenyAllIngress"
    DENY_ALL_EGRESS = "DenyAllEgress"


# ---------------------------------------------------------------------------
# spec sub-objects
# ---------------------------------------------------------------------------

class ScaleStrategy(K8sModel):
    """raycluster_types.go:489-492."""

    workers_to_delete: Optional[List[str]] = None


class HeadGroupSpec(K8sModel):
    """raycluster_types.go:375-403."""

    template: PodTemplateSpec = Field(default_factory=PodTemplateSpec)
    head_service: Optional[Service] = None
    enable_ingress: Optional[bool] = None
    ingress_options: Optional[Dict[str, Any]] = None
    resources: Optional[Dict[str, str]] = None
    labels: Optional[Dict[str, str]] = None
    ray_start_params: Dict[str, str] = Field(default_factory=dict)
    service_type: Optional[str] = None


class WorkerGroupSpec(K8sModel):
    """raycluster_types.go:436-486."""

    suspend: Optional[bool] = None
    group_name: str = ""
    replicas: Optional[int] = None
    min_replicas: Optional[int] = 0
    max_replicas: Optional[int] = 2147483647
    idle_timeout_seconds: Optional[int] = None
    priority: Optional[int] = None
    resources: Optional[Dict[str, str]] = None
    labels: Optional[Dict[str, str]] = None
    ray_start_params: Dict[str, str] = Field(default_factory=dict)
    template: PodTemplateSpec = Field(default_factory=PodTemplateSpec)
    scale_strategy: ScaleStrategy = Field(default_factory=ScaleStrategy)
    num_of_hosts: int = 1


class AutoscalerOptions(K8sModel):
    """raycluster_types.go:495-544."""

    resources: Optional[ResourceRequirements] = None
    image: Optional[str] = None
    image_pull_policy: Optional[str] = None
    security_context: Optional[SecurityContext] = None
    idle_timeout_seconds: Optional[int] = None
    upscaling_mode: Optional[str] = None
    version: Optional[str] = None
    env: Optional[List[EnvVar]] = None
    env_from: Optional[List[Dict[str, Any]]] = None
    volume_mounts: Optional[List[VolumeMount]] = None
    command: Optional[List[str]] = None
    args: Optional[List[str]] = None


class RedisCredential(K8sModel):
    value_from: Optional[Dict[str, Any]] = None
    value: Optional[str] = None


class GcsEmbeddedStorage(K8sModel):
    """raycluster_types.go:188-246 (RocksDB-on-PVC backend)."""

    claim_name: Optional[str] = None
    size: Optional[str] = None
    storage_class_name: Optional[str] = None
    access_modes: Optional[List[str]] = None
    sub_path: Optional[str] = None
    deletion_policy: Optional[str] = None  # Retain | Delete


class GcsFaultToleranceOptions(K8sModel):
    """raycluster_types.go:152-180."""

    backend: Optional[str] = None  # redis | embedded
    redis_username: Optional[RedisCredential] = None
    redis_password: Optional[RedisCredential] = None
    external_storage_namespace: Optional[str] = None
    redis_address: Optional[str] = None
    storage: Optional[GcsEmbeddedStorage] = None


class AuthOptions(K8sModel):
    """raycluster_types.go:112-137."""

    enable_k8s_token_auth: Optional[bool] = Field(
        default=None, alias="enableK8sTokenAuth")
    secret_name: Optional[str] = None
    mode: Optional[str] = None  # token | disabled


class TLSOptions(K8sModel):
    """raycluster_types.go:78-83."""

    enabled: Optional[bool] = None


class RayClusterUpgradeStrategy(K8sModel):
    type: Optional[str] = None  # Recreate | None


class NetworkPolicyRules(K8sModel):
    ingress_rules: Optional[List[Dict[str, Any]]] = None
    egress_rules: Optional[List[Dict[str, Any]]] = None


class WorkerGroupNetworkPolicyRules(NetworkPolicyRules):
    group_name: str = ""


class NetworkPolicyConfig(K8sModel):
    """raycluster_types.go:314-346."""

    mode: Optional[str] = None  # DenyAll | DenyAllIngress | DenyAllEgress
    head: Optional[NetworkPolicyRules] = None
    worker: Optional[NetworkPolicyRules] = None
    worker_groups: Optional[List[WorkerGroupNetworkPolicyRules]] = None


class CollectorOptions(K8sModel):
    """History-server collector sidecar options (raycluster_types.go:277-293)."""

    image: Optional[str] = None
    image_pull_policy: Optional[str] = None
    resources: Optional[ResourceRequirements] = None
    env: Optional[List[EnvVar]] = None


class HistoryServerOptions(K8sModel):
    collector_options: Optional[CollectorOptions] = None


class RayClusterSpec(K8sModel):
    """raycluster_types.go:14-71."""

    upgrade_strategy: Optional[RayClusterUpgradeStrategy] = None
    auth_options: Optional[AuthOptions] = None
    suspend: Optional[bool] = None
    managed_by: Optional[str] = None
    autoscaler_options: Optional[AutoscalerOptions] = None
    head_service_annotations: Optional[Dict[str, str]] = None
    enable_in_tree_autoscaling: Optional[bool] = None
    gcs_fault_tolerance_options: Optional[GcsFaultToleranceOptions] = None
    history_server_options: Optional[HistoryServerOptions] = None
    network_policy: Optional[NetworkPolicyConfig] = None
    tls_options: Optional[TLSOptions] = None
    head_group_spec: HeadGroupSpec = Field(default_factory=HeadGroupSpec)
    ray_version: Optional[str] = None
    worker_group_specs: List[WorkerGroupSpec] = Field(default_factory=list)


# ---------------------------------------------------------------------------
# status
# ---------------------------------------------------------------------------

class HeadInfo(K8sModel):
    pod_ip: Optional[str] = Field(default=None, alias="podIP")
    service_ip: Optional[str] = Field(default=None, alias="serviceIP")
    pod_name: Optional[str] = None
    service_name: Optional[str] = None


class RayClusterStatus(K8sModel):
    """raycluster_types.go:568-639 (no desiredTPU — MI355X scoping)."""

    state: Optional[str] = None
    desired_cpu: Optional[str] = Field(default=None, alias="desiredCPU")
    desired_memory: Optional[str] = None
    desired_gpu: Optional[str] = Field(default=None, alias="desiredGPU")
    last_update_time: Optional[str] = None
    state_transition_times: Optional[Dict[str, str]] = None
    endpoints: Optional[Dict[str, str]] = None
    head: HeadInfo = Field(default_factory=HeadInfo)
    reason: Optional[str] = None
    conditions: Optional[List[Condition]] = None
    ready_worker_replicas: int = 0
    available_worker_replicas: int = 0
    desired_worker_replicas: int = 0
    min_worker_replicas: int = 0
    max_worker_replicas: int = 0
    observed_generation: Optional[int] = None


class RayCluster(K8sModel):
    api_version: str = C.API_VERSION
    kind: str = C.KIND_RAYCLUSTER
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: RayClusterSpec = Field(default_factory=RayClusterSpec)
    status: RayClusterStatus = Field(default_factory=RayClusterStatus)

    # -- convenience ---------------------------------------------------
    def get_condition(self, cond_type: str) -> Optional[Condition]:
        for c in self.status.conditions or []:
            if c.type == cond_type:
                return c
        return None

    def condition_true(self, cond_type: str) -> bool:
        c = self.get_condition(cond_type)
        return c is not None and c.status == "True"
