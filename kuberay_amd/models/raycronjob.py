"""ray.io/v1 RayCronJob types (reference: ray-operator/apis/ray/v1/raycronjob_types.go)."""
from __future__ import annotations

from typing import Optional

from pydantic import Field

from ..kube.objects import K8sModel, ObjectMeta
from ..utils import constants as C
from .rayjob import RayJobSpec


class RayCronJobSpec(K8sModel):
    """raycronjob_types.go:10-25."""

    job_template: RayJobSpec = Field(default_factory=RayJobSpec)
    schedule: str = ""
    time_zone: Optional[str] = None
    suspend: Optional[bool] = None


class RayCronJobStatus(K8sModel):
    last_schedule_time: Optional[str] = None


class RayCronJob(K8sModel):
    api_version: str = C.API_VERSION
    kind: str = C.KIND_RAYCRONJOB
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: RayCronJobSpec = Field(default_factory=RayCronJobSpec)
    status: RayCronJobStatus = Field(default_factory=RayCronJobStatus)
