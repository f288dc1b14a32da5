"""Spec validation for all four CRDs
(reference: ray-operator/controllers/ray/utils/validation.go).

Pure functions returning a list of error strings (empty = valid). The same
functions back the reconcilers' early validation, the validating webhooks,
and the apiserver.
"""
from __future__ import annotations

import re
from typing import List

from ..kube.objects import ObjectMeta
from ..models.raycluster import RayCluster, RayClusterSpec
from ..models.raycronjob import RayCronJob
from ..models.rayjob import (
    DeletionPolicyType,
    JobDeploymentStatus,
    JobStatus,
    JobSubmissionMode,
    RayJob,
)
from ..models.rayservice import RayService, RayServiceUpgradeType
from . import constants as C

_NAME_RE = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")


def validate_raycluster_metadata(meta: ObjectMeta) -> List[str]:
    """validation.go ValidateRayClusterMetadata."""
    errs = []
    name = meta.name or ""
    if len(name) > 63:
        errs.append(f"RayCluster name '{name}' exceeds 63 characters")
    if name and not _NAME_RE.match(name):
        errs.append(f"RayCluster name '{name}' is not a valid DNS-1035 label")
    return errs


def validate_raycluster_spec(cluster: RayCluster) -> List[str]:
    """validation.go:103 ValidateRayClusterSpec."""
    errs = []
    spec = cluster.spec
    if not spec.head_group_spec.template.spec.containers:
        errs.append("headGroupSpec should have at least one container")
    group_names = set()
    for group in spec.worker_group_specs:
        if not group.group_name:
            errs.append("workerGroupSpec groupName must not be empty")
        if group.group_name in group_names:
            errs.append(f"duplicate worker group name '{group.group_name}'")
        group_names.add(group.group_name)
        if not group.template.spec.containers:
            errs.append(f"worker group '{group.group_name}' should have at least one container")
        min_r = group.min_replicas or 0
        max_r = group.max_replicas if group.max_replicas is not None else 2**31 - 1
        if min_r > max_r:
            errs.append(
                f"worker group '{group.group_name}': minReplicas {min_r} > maxReplicas {max_r}")
        if group.replicas is not None and group.replicas < 0:
            errs.append(f"worker group '{group.group_name}': replicas must be >= 0")
        if group.num_of_hosts < 0:
            errs.append(f"worker group '{group.group_name}': numOfHosts must be >= 0")
        if group.idle_timeout_seconds is not None and not _autoscaler_v2(spec):
            errs.append(
                f"worker group '{group.group_name}': idleTimeoutSeconds requires "
                "autoscaler v2 (spec.autoscalerOptions.version: v2)")
        errs += _validate_group_resources_and_labels(
            group.group_name or "worker", group.ray_start_params,
            getattr(group, "resources", None), getattr(group, "labels", None))
    errs += _validate_group_resources_and_labels(
        "head", spec.head_group_spec.ray_start_params,
        getattr(spec.head_group_spec, "resources", None),
        getattr(spec.head_group_spec, "labels", None))
    errs += _validate_gcs_ft(cluster)
    errs += _validate_auth(spec)
    errs += _validate_network_policy(spec)
    errs += _validate_tls(spec)
    errs += _validate_cluster_upgrade(cluster)
    errs += _validate_collector_options(spec)
    for group in spec.worker_group_specs:
        errs += _validate_worker_priority(group, spec)
    return errs


# env names injected into the collector by the operator; user overrides
# would silently fight the injection (validation.go:1029-1064 analog)
_COLLECTOR_MANAGED_ENV = {"POD_IP", "RAY_ROLE", "OWNER_KIND", "OWNER_NAME",
                          C.RAY_CLUSTER_NAMESPACE, "EVENTS_PORT"}
_COLLECTOR_BACKEND_REQUIRES = {
    "s3": "S3_REGION", "gcs": "GCS_BUCKET", "azure": "AZURE_STORAGE_ACCOUNT",
    "oss": "OSS_ENDPOINT"}


def _validate_collector_options(spec: RayClusterSpec) -> List[str]:
    errs: List[str] = []
    hso = getattr(spec, "history_server_options", None)
    opts = getattr(hso, "collector_options", None) if hso else None
    if opts is None:
        return errs
    env = {e.name: e for e in (getattr(opts, "env", None) or [])}
    for name in sorted(_COLLECTOR_MANAGED_ENV & set(env)):
        errs.append(
            f"historyServerOptions.collectorOptions.env must not contain "
            f"{name}: it is injected by the operator")
    backend = env.get("STORAGE_BACKEND")
    if backend is not None and backend.value:
        needed = _COLLECTOR_BACKEND_REQUIRES.get(backend.value.lower())
        if needed and not (env.get(needed) and
                           (env[needed].value or env[needed].value_from)):
            errs.append(
                f"{needed} env must be set when STORAGE_BACKEND is "
                f"{backend.value}")
    return errs


def _validate_cluster_upgrade(cluster: RayCluster) -> List[str]:
    """validation.go:42-58 ValidateRayClusterUpgradeOptions."""
    errs: List[str] = []
    us = cluster.spec.upgrade_strategy
    if us is None or us.type is None:
        return errs
    if us.type not in ("Recreate", "None"):
        errs.append(f"upgradeStrategy.type '{us.type}' is invalid; valid "
                    "options are Recreate or None")
    creator = (cluster.metadata.labels or {}).get(
        C.RAY_ORIGINATED_FROM_CRD_LABEL_KEY)
    if creator in ("RayJob", "RayService"):
        errs.append(f"upgradeStrategy cannot be set when RayCluster is "
                    f"created by {creator}")
    return errs


def _validate_worker_priority(group, spec: RayClusterSpec) -> List[str]:
    """validation.go:1113-1142 — priority needs Ray >= 2.56 + autoscaler v2."""
    errs: List[str] = []
    if not group.priority:
        return errs
    name = group.group_name or "worker"
    if not spec.ray_version:
        errs.append(f"worker group {name}: priority is set, but rayVersion "
                    "was not specified (Ray >= 2.56.0 required)")
        return errs
    try:
        parts = tuple(int(x) for x in
                      re.match(r"(\d+)\.(\d+)", spec.ray_version).groups())
    except (AttributeError, ValueError):
        errs.append(f"worker group {name}: priority is set, but rayVersion "
                    f"format is invalid: {spec.ray_version}")
        return errs
    if parts < (2, 56):
        errs.append(f"worker group {name}: priority requires Ray >= 2.56.0, "
                    f"got {spec.ray_version}")
    if not _autoscaler_v2(spec):
        head_env = []
        if spec.head_group_spec.template.spec.containers:
            head_env = spec.head_group_spec.template.spec.containers[0].env or []
        if not any(e.name == C.RAY_ENABLE_AUTOSCALER_V2
                   and str(e.value).lower() in ("1", "true")
                   for e in head_env):
            errs.append(f"worker group {name}: priority is only supported "
                        "with autoscaler v2 enabled")
    return errs


def _validate_tls(spec: RayClusterSpec) -> List[str]:
    """validation.go:446-480 validateTLSOptions — with tlsOptions enabled,
    the operator owns the TLS env vars and cert mount; user-set values
    would silently fight it."""
    errs: List[str] = []
    opts = spec.tls_options
    enabled = (opts.get("enabled") if isinstance(opts, dict)
               else getattr(opts, "enabled", None)) if opts else None
    if not enabled:
        return errs
    forbidden = {C.RAY_USE_TLS, C.RAY_TLS_SERVER_CERT, C.RAY_TLS_SERVER_KEY,
                 C.RAY_TLS_CA_CERT}

    def check(template, where):
        for container in template.spec.containers or []:
            for env in container.env or []:
                if env.name in forbidden:
                    errs.append(
                        f"cannot set {env.name} env in {where} when "
                        "tlsOptions is enabled — the operator manages TLS")
            for vm in container.volume_mounts or []:
                if vm.mount_path == C.RAY_TLS_CERT_MOUNT_PATH or \
                        vm.name == C.RAY_TLS_VOLUME_NAME:
                    errs.append(
                        f"cannot mount {vm.mount_path} in {where} when "
                        "tlsOptions is enabled — the operator manages the "
                        "cert mount")

    check(spec.head_group_spec.template, "head Pod")
    for group in spec.worker_group_specs:
        check(group.template, f"worker group '{group.group_name}'")
    return errs


_LABEL_KEY_RE = re.compile(
    r"^([a-z0-9]([-a-z0-9.]*[a-z0-9])?/)?[A-Za-z0-9]([-A-Za-z0-9_.]*[A-Za-z0-9])?$")
_LABEL_VALUE_RE = re.compile(r"^([A-Za-z0-9]([-A-Za-z0-9_.]*[A-Za-z0-9])?)?$")


def _validate_group_resources_and_labels(group_name: str, ray_start_params,
                                         resources, labels) -> List[str]:
    """validation.go:60-101 — the top-level group `resources`/`labels`
    fields own their rayStartParams keys; manual duplicates conflict."""
    errs: List[str] = []
    params = ray_start_params or {}
    has_param_resources = any(params.get(k) for k in
                              ("num-cpus", "num-gpus", "memory", "resources"))
    if has_param_resources and resources:
        errs.append(
            f"resource fields should not be set in both rayStartParams and "
            f"resources for {group_name} group; please use only one")
    if "labels" in params:
        errs.append(
            f"rayStartParams['labels'] is not supported for {group_name} "
            "group; please use the top-level labels field instead")
    for key, val in (labels or {}).items():
        if len(key) > 253 or not _LABEL_KEY_RE.match(key):
            errs.append(f"invalid label key for {group_name} group: '{key}'")
        if len(str(val)) > 63 or not _LABEL_VALUE_RE.match(str(val)):
            errs.append(
                f"invalid label value for key '{key}' in {group_name} "
                f"group: '{val}'")
    return errs


def _autoscaler_v2(spec: RayClusterSpec) -> bool:
    return bool(spec.autoscaler_options and spec.autoscaler_options.version == "v2")


def _validate_gcs_ft(cluster: RayCluster) -> List[str]:
    """validation.go:333 (GCS FT backend rules)."""
    errs = []
    spec = cluster.spec
    opts = spec.gcs_fault_tolerance_options
    annotations = cluster.metadata.annotations or {}
    ft_annotation = annotations.get(C.RAY_FT_ENABLED_ANNOTATION_KEY)
    if opts is not None and ft_annotation is not None:
        errs.append(
            f"annotation {C.RAY_FT_ENABLED_ANNOTATION_KEY} and "
            "gcsFaultToleranceOptions are mutually exclusive")
    if opts is not None:
        backend = opts.backend or ("redis" if opts.redis_address else None)
        if backend == "embedded":
            if opts.redis_address or opts.redis_username or opts.redis_password:
                errs.append("embedded GCS storage backend cannot set redis fields")
        elif backend == "redis":
            if not opts.redis_address:
                errs.append("redis GCS backend requires redisAddress")
        if opts.storage is not None and backend != "embedded":
            errs.append("gcsFaultToleranceOptions.storage requires backend: embedded")
    head_container = (spec.head_group_spec.template.spec.containers[0]
                      if spec.head_group_spec.template.spec.containers else None)
    if head_container is not None and opts is not None:
        for env in head_container.env or []:
            if env.name == C.RAY_REDIS_ADDRESS and opts.backend == "embedded":
                errs.append("RAY_REDIS_ADDRESS env is invalid with embedded GCS backend")
    return errs


def _validate_auth(spec: RayClusterSpec) -> List[str]:
    errs = []
    ao = spec.auth_options
    if ao is None:
        return errs
    if ao.mode not in (None, "token", "disabled"):
        errs.append(f"authOptions.mode must be 'token' or 'disabled', got '{ao.mode}'")
    return errs


def _validate_network_policy(spec: RayClusterSpec) -> List[str]:
    """validation.go:380."""
    errs = []
    np = spec.network_policy
    if np is None:
        return errs
    if np.mode not in (None, "DenyAll", "DenyAllIngress", "DenyAllEgress"):
        errs.append(f"networkPolicy.mode invalid: '{np.mode}'")
    group_names = {g.group_name for g in spec.worker_group_specs}
    for wg in np.worker_groups or []:
        if wg.group_name not in group_names:
            errs.append(f"networkPolicy.workerGroups references unknown group '{wg.group_name}'")
    return errs


# ---------------------------------------------------------------------------
# RayJob (validation.go:543-967)
# ---------------------------------------------------------------------------

def validate_rayjob_metadata(meta: ObjectMeta) -> List[str]:
    errs = []
    name = meta.name or ""
    if len(name) > 63:
        errs.append(f"RayJob name '{name}' exceeds 63 characters")
    return errs


def validate_rayjob_spec(rayjob: RayJob) -> List[str]:
    errs = []
    spec = rayjob.spec
    if spec.submission_mode not in JobSubmissionMode.ALL:
        errs.append(f"invalid submissionMode '{spec.submission_mode}'")
    if spec.ray_cluster_spec is None and not spec.cluster_selector:
        errs.append("one of rayClusterSpec or clusterSelector must be set")
    if spec.ray_cluster_spec is not None and spec.cluster_selector:
        errs.append("rayClusterSpec and clusterSelector are mutually exclusive")
    if spec.submission_mode == JobSubmissionMode.INTERACTIVE and spec.entrypoint:
        errs.append("entrypoint must be empty in InteractiveMode")
    if spec.submission_mode != JobSubmissionMode.INTERACTIVE and not spec.entrypoint:
        errs.append("entrypoint is required unless submissionMode is InteractiveMode")
    if spec.submission_mode == JobSubmissionMode.SIDECAR and spec.cluster_selector:
        errs.append("SidecarMode requires an operator-managed cluster (no clusterSelector)")
    if spec.submitter_pod_template is not None and spec.submission_mode not in (
            JobSubmissionMode.K8S_JOB,):
        errs.append("submitterPodTemplate only applies to K8sJobMode")
    if spec.backoff_limit is not None and spec.backoff_limit < 0:
        errs.append("backoffLimit must be >= 0")
    if spec.active_deadline_seconds is not None and spec.active_deadline_seconds <= 0:
        errs.append("activeDeadlineSeconds must be > 0")
    if spec.ttl_seconds_after_finished < 0:
        errs.append("ttlSecondsAfterFinished must be >= 0")
    if spec.ttl_seconds_after_finished > 0 and not spec.shutdown_after_job_finishes:
        errs.append("ttlSecondsAfterFinished requires shutdownAfterJobFinishes: true")
    if spec.shutdown_after_job_finishes and spec.cluster_selector:
        errs.append("shutdownAfterJobFinishes is invalid with clusterSelector "
                    "(the job does not own the cluster)")
    errs += _validate_deletion_strategy(rayjob)
    if spec.ray_cluster_spec is not None:
        sub = RayCluster(metadata=rayjob.metadata, spec=spec.ray_cluster_spec)
        errs += validate_raycluster_spec(sub)
    return errs


def _validate_deletion_strategy(rayjob: RayJob) -> List[str]:
    errs = []
    ds = rayjob.spec.deletion_strategy
    if ds is None:
        return errs
    legacy = ds.on_success is not None or ds.on_failure is not None
    rules = ds.deletion_rules is not None
    if legacy and rules:
        errs.append("deletionStrategy: onSuccess/onFailure and deletionRules are mutually exclusive")
    if legacy and (ds.on_success is None or ds.on_failure is None):
        errs.append("deletionStrategy: both onSuccess and onFailure must be set")
    for block in (ds.on_success, ds.on_failure):
        if block is not None and block.policy not in DeletionPolicyType.ALL:
            errs.append(f"deletionStrategy: invalid policy '{block.policy}'")
    for rule in ds.deletion_rules or []:
        if rule.policy not in DeletionPolicyType.ALL:
            errs.append(f"deletionRules: invalid policy '{rule.policy}'")
        if rule.condition.ttl_seconds < 0:
            errs.append("deletionRules: ttlSeconds must be >= 0")
        js = rule.condition.job_status
        if js is not None and js not in JobStatus.TERMINAL:
            errs.append(f"deletionRules: jobStatus '{js}' is not terminal")
        jds = rule.condition.job_deployment_status
        if jds is not None and jds not in (JobDeploymentStatus.COMPLETE,
                                           JobDeploymentStatus.FAILED):
            errs.append(f"deletionRules: jobDeploymentStatus '{jds}' is not terminal")
    if (rayjob.spec.shutdown_after_job_finishes and rules):
        errs.append("deletionRules and shutdownAfterJobFinishes are mutually exclusive")
    return errs


# ---------------------------------------------------------------------------
# RayService (validation.go:680)
# ---------------------------------------------------------------------------

def validate_rayservice_metadata(meta: ObjectMeta) -> List[str]:
    errs = []
    name = meta.name or ""
    if len(name) > 63:
        errs.append(f"RayService name '{name}' exceeds 63 characters")
    return errs


def _validate_initializing_timeout(annotations) -> List[str]:
    """validation.go:651-678 — ray.io/initializing-timeout accepts a Go-style
    duration ("30m", "1h") or positive integer seconds."""
    value = (annotations or {}).get(
        C.RAY_SERVICE_INITIALIZING_TIMEOUT_ANNOTATION)
    if not value:
        return []
    m = re.fullmatch(r"(\d+(\.\d+)?)(s|m|h)?", value)
    if m is None:
        return [f"annotation {C.RAY_SERVICE_INITIALIZING_TIMEOUT_ANNOTATION} "
                f"has invalid format: {value!r} (expected a duration like "
                "'5m'/'1h' or positive integer seconds)"]
    if float(m.group(1)) <= 0:
        return [f"annotation {C.RAY_SERVICE_INITIALIZING_TIMEOUT_ANNOTATION} "
                f"must be positive, got: {value!r}"]
    return []


def validate_rayservice_spec(rayservice: RayService) -> List[str]:
    errs = []
    spec = rayservice.spec
    errs += _validate_initializing_timeout(rayservice.metadata.annotations)
    sub = RayCluster(metadata=rayservice.metadata, spec=spec.ray_cluster_spec)
    errs += validate_raycluster_spec(sub)
    us = spec.upgrade_strategy
    if us is not None and us.type not in (
            None, RayServiceUpgradeType.NEW_CLUSTER,
            RayServiceUpgradeType.NEW_CLUSTER_WITH_INCREMENTAL_UPGRADE,
            RayServiceUpgradeType.NONE):
        errs.append(f"invalid upgradeStrategy.type '{us.type}'")
    if us is not None and us.type == RayServiceUpgradeType.NEW_CLUSTER_WITH_INCREMENTAL_UPGRADE:
        opts = us.cluster_upgrade_options
        if opts is None:
            errs.append("NewClusterWithIncrementalUpgrade requires clusterUpgradeOptions")
        else:
            if opts.gateway_class_name in (None, ""):
                errs.append("clusterUpgradeOptions.gatewayClassName is required")
            for field, val in (("stepSizePercent", opts.step_size_percent),
                               ("intervalSeconds", opts.interval_seconds)):
                if val is None:
                    errs.append(f"clusterUpgradeOptions.{field} is required")
                elif val < 0 or (field == "stepSizePercent" and val > 100):
                    errs.append(f"clusterUpgradeOptions.{field} out of range")
            if opts.max_surge_percent is not None and not (0 <= opts.max_surge_percent <= 100):
                errs.append("clusterUpgradeOptions.maxSurgePercent must be in [0,100]")
    if spec.serve_config_v2:
        import yaml
        try:
            data = yaml.safe_load(spec.serve_config_v2)
            if not isinstance(data, dict) or "applications" not in data:
                errs.append("serveConfigV2 must be a YAML map with an 'applications' list")
        except yaml.YAMLError as e:
            errs.append(f"serveConfigV2 is not valid YAML: {e}")
    return errs


# ---------------------------------------------------------------------------
# RayCronJob (validation.go:969)
# ---------------------------------------------------------------------------

def validate_raycronjob_spec(cronjob: RayCronJob) -> List[str]:
    errs = []
    from .cron import parse_cron
    if not cronjob.spec.schedule:
        errs.append("schedule is required")
    else:
        try:
            parse_cron(cronjob.spec.schedule)
        except ValueError as e:
            errs.append(f"invalid schedule: {e}")
    if cronjob.spec.time_zone:
        try:
            from zoneinfo import ZoneInfo
            ZoneInfo(cronjob.spec.time_zone)
        except Exception:
            errs.append(f"invalid timeZone {cronjob.spec.time_zone!r}")
    job = RayJob(metadata=cronjob.metadata, spec=cronjob.spec.job_template)
    errs += validate_rayjob_spec(job)
    return errs
