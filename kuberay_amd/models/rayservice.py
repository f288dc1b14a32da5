"""ray.io/v1 RayService types (reference: ray-operator/apis/ray/v1/rayservice_types.go)."""
from __future__ import annotations

from typing import Dict, List, Optional

from pydantic import Field

from ..kube.objects import Condition, K8sModel, ObjectMeta, Service
from ..utils import constants as C
from .raycluster import RayClusterSpec, RayClusterStatus


class RayServiceUpgradeType:
    NEW_CLUSTER_WITH_INCREMENTAL_UPGRADE = "NewClusterWithIncrementalUpgrade"
    NEW_CLUSTER = "NewCluster"
    NONE = "None"


class ServiceStatus:
    """rayservice_types.go ServiceStatus values."""

    RUNNING = "Running"
    NOT_RUNNING = "NotRunning"


class ApplicationStatus:
    """Mirror of Ray Serve's ApplicationStatus enum (serve.proto)."""

    NOT_STARTED = "NOT_STARTED"
    DEPLOYING = "DEPLOYING"
    RUNNING = "RUNNING"
    DEPLOY_FAILED = "DEPLOY_FAILED"
    DELETING = "DELETING"
    UNHEALTHY = "UNHEALTHY"

    HEALTHY = {RUNNING}
    TERMINAL_BAD = {DEPLOY_FAILED, UNHEALTHY}


class RayServiceConditionType:
    READY = "Ready"
    UPGRADE_IN_PROGRESS = "UpgradeInProgress"
    ROLLBACK_IN_PROGRESS = "RollbackInProgress"
    SUSPENDING = "Suspending"
    SUSPENDED = "Suspended"


class RayServiceConditionReason:
    NON_ZERO_SERVE_ENDPOINTS = "NonZeroServeEndpoints"
    ZERO_SERVE_ENDPOINTS = "ZeroServeEndpoints"
    INITIALIZING_TIMEOUT = "InitializingTimeout"


class ClusterUpgradeOptions(K8sModel):
    """Incremental-upgrade knobs (rayservice_types.go:64-76)."""

    max_surge_percent: Optional[int] = None
    step_size_percent: Optional[int] = None
    interval_seconds: Optional[int] = None
    gateway_class_name: Optional[str] = None


class RayServiceUpgradeStrategy(K8sModel):
    type: Optional[str] = None
    cluster_upgrade_options: Optional[ClusterUpgradeOptions] = None


class RayServiceSpec(K8sModel):
    """rayservice_types.go:88-130."""

    ray_cluster_deletion_delay_seconds: Optional[int] = None
    service_unhealthy_second_threshold: Optional[int] = None
    deployment_unhealthy_second_threshold: Optional[int] = None
    serve_service: Optional[Service] = None
    upgrade_strategy: Optional[RayServiceUpgradeStrategy] = None
    managed_by: Optional[str] = None
    serve_config_v2: Optional[str] = Field(default=None, alias="serveConfigV2")
    ray_cluster_spec: RayClusterSpec = Field(
        default_factory=RayClusterSpec, alias="rayClusterConfig"
    )
    exclude_head_pod_from_serve_svc: bool = False
    suspend: bool = False


class ServeDeploymentStatus(K8sModel):
    status: Optional[str] = None
    message: Optional[str] = None


class AppStatus(K8sModel):
    deployments: Optional[Dict[str, ServeDeploymentStatus]] = Field(
        default=None, alias="serveDeploymentStatuses"
    )
    status: Optional[str] = None
    message: Optional[str] = None


class RayServiceStatus(K8sModel):
    """Per-cluster (active or pending) status block (rayservice_types.go:165-190)."""

    applications: Optional[Dict[str, AppStatus]] = Field(
        default=None, alias="applicationStatuses"
    )
    target_capacity: Optional[int] = None
    traffic_routed_percent: Optional[int] = None
    last_traffic_migrated_time: Optional[str] = None
    ray_cluster_name: Optional[str] = None
    ray_cluster_status: RayClusterStatus = Field(default_factory=RayClusterStatus)


class RayServiceStatuses(K8sModel):
    """Top-level status (rayservice_types.go:133-161)."""

    conditions: Optional[List[Condition]] = None
    last_update_time: Optional[str] = None
    service_status: Optional[str] = None
    active_service_status: RayServiceStatus = Field(default_factory=RayServiceStatus)
    pending_service_status: RayServiceStatus = Field(default_factory=RayServiceStatus)
    num_serve_endpoints: int = 0
    observed_generation: Optional[int] = None


class RayService(K8sModel):
    api_version: str = C.API_VERSION
    kind: str = C.KIND_RAYSERVICE
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: RayServiceSpec = Field(default_factory=RayServiceSpec)
    status: RayServiceStatuses = Field(default_factory=RayServiceStatuses)

    def get_condition(self, cond_type: str) -> Optional[Condition]:
        for c in self.status.conditions or []:
            if c.type == cond_type:
                return c
        return None

    def condition_true(self, cond_type: str) -> bool:
        c = self.get_condition(cond_type)
        return c is not None and c.status == "True"
