"""ray.io/v1 RayJob types (reference: ray-operator/apis/ray/v1/rayjob_types.go)."""
from __future__ import annotations

from typing import Dict, List, Optional

from pydantic import Field

from ..kube.objects import Condition, K8sModel, ObjectMeta, PodTemplateSpec
from ..utils import constants as C
from .raycluster import RayClusterSpec, RayClusterStatus


class JobStatus:
    """Ray application job status (rayjob_types.go:17-24)."""

    NEW = ""
    PENDING = "PENDING"
    RUNNING = "RUNNING"
    STOPPED = "STOPPED"
    SUCCEEDED = "SUCCEEDED"
    FAILED = "FAILED"

    TERMINAL = {STOPPED, SUCCEEDED, FAILED}

    @classmethod
    def is_terminal(cls, s: Optional[str]) -> bool:
        return s in cls.TERMINAL


class JobDeploymentStatus:
    """Operator-level lifecycle state (rayjob_types.go:48-59)."""

    NEW = ""
    INITIALIZING = "Initializing"
    RUNNING = "Running"
    COMPLETE = "Complete"
    FAILED = "Failed"
    VALIDATION_FAILED = "ValidationFailed"
    SUSPENDING = "Suspending"
    SUSPENDED = "Suspended"
    RETRYING = "Retrying"
    WAITING = "Waiting"

    TERMINAL = {COMPLETE, FAILED, VALIDATION_FAILED}


class JobFailedReason:
    SUBMISSION_FAILED = "SubmissionFailed"
    DEADLINE_EXCEEDED = "DeadlineExceeded"
    PRE_RUNNING_DEADLINE_EXCEEDED = "PreRunningDeadlineExceeded"
    APP_FAILED = "AppFailed"
    TRANSITION_GRACE_PERIOD_EXCEEDED = "JobDeploymentStatusTransitionGracePeriodExceeded"
    JOB_STATUS_CHECK_TIMEOUT_EXCEEDED = "JobStatusCheckTimeoutExceeded"
    VALIDATION_FAILED = "ValidationFailed"


class JobSubmissionMode:
    K8S_JOB = "K8sJobMode"
    HTTP = "HTTPMode"
    INTERACTIVE = "InteractiveMode"
    SIDECAR = "SidecarMode"

    ALL = {K8S_JOB, HTTP, INTERACTIVE, SIDECAR}


class DeletionPolicyType:
    DELETE_CLUSTER = "DeleteCluster"
    DELETE_WORKERS = "DeleteWorkers"
    DELETE_SELF = "DeleteSelf"
    DELETE_NONE = "DeleteNone"

    ALL = {DELETE_CLUSTER, DELETE_WORKERS, DELETE_SELF, DELETE_NONE}


class DeletionPolicy(K8sModel):
    policy: Optional[str] = None


class DeletionCondition(K8sModel):
    """rayjob_types.go:146-168."""

    job_status: Optional[str] = None
    job_deployment_status: Optional[str] = None
    ttl_seconds: int = 0


class DeletionRule(K8sModel):
    policy: str = DeletionPolicyType.DELETE_NONE
    condition: DeletionCondition = Field(default_factory=DeletionCondition)


class DeletionStrategy(K8sModel):
    """rayjob_types.go:108-128 (legacy onSuccess/onFailure + new deletionRules)."""

    on_success: Optional[DeletionPolicy] = None
    on_failure: Optional[DeletionPolicy] = None
    deletion_rules: Optional[List[DeletionRule]] = None


class SubmitterConfig(K8sModel):
    backoff_limit: Optional[int] = None


class RayJobSpec(K8sModel):
    """rayjob_types.go:208-301."""

    active_deadline_seconds: Optional[int] = None
    backoff_limit: Optional[int] = None
    ray_cluster_spec: Optional[RayClusterSpec] = None
    submitter_pod_template: Optional[PodTemplateSpec] = None
    metadata: Optional[Dict[str, str]] = None
    cluster_selector: Optional[Dict[str, str]] = None
    submitter_config: Optional[SubmitterConfig] = None
    managed_by: Optional[str] = None
    deletion_strategy: Optional[DeletionStrategy] = None
    entrypoint: Optional[str] = None
    runtime_env_yaml: Optional[str] = Field(default=None, alias="runtimeEnvYAML")
    job_id: Optional[str] = None
    submission_mode: str = JobSubmissionMode.K8S_JOB
    entrypoint_resources: Optional[str] = None
    entrypoint_num_cpus: Optional[float] = None
    entrypoint_num_gpus: Optional[float] = None
    ttl_seconds_after_finished: int = 0
    pre_running_deadline_seconds: Optional[int] = None
    shutdown_after_job_finishes: bool = False
    suspend: bool = False


class RayJobStatusInfo(K8sModel):
    start_time: Optional[str] = None
    end_time: Optional[str] = None


class RayJobStatus(K8sModel):
    """rayjob_types.go:304-352."""

    ray_job_info: RayJobStatusInfo = Field(
        default_factory=RayJobStatusInfo, alias="rayJobInfo"
    )
    job_id: Optional[str] = None
    ray_cluster_name: Optional[str] = None
    dashboard_url: Optional[str] = Field(default=None, alias="dashboardURL")
    job_status: str = JobStatus.NEW
    job_deployment_status: str = JobDeploymentStatus.NEW
    reason: Optional[str] = None
    message: Optional[str] = None
    start_time: Optional[str] = None
    end_time: Optional[str] = None
    succeeded: Optional[int] = None
    failed: Optional[int] = None
    ray_cluster_status: RayClusterStatus = Field(default_factory=RayClusterStatus)
    job_status_check_failure_start_time: Optional[str] = None
    observed_generation: Optional[int] = None
    conditions: Optional[List[Condition]] = None


class RayJob(K8sModel):
    api_version: str = C.API_VERSION
    kind: str = C.KIND_RAYJOB
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: RayJobSpec = Field(default_factory=RayJobSpec)
    status: RayJobStatus = Field(default_factory=RayJobStatus)
