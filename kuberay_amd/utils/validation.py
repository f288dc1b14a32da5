"""Spec validation for all four CRDs
(reference: ray-operator/controllers/ray/utils/validation.go:23-1142).

Pure functions returning a list of error strings (empty = valid). The same
functions back the reconcilers' early validation, the validating webhooks,
and the apiserver. Rule-by-rule parity with the reference file is exercised
by tests/test_validation_parity.py (one table row per reference branch).
"""
from __future__ import annotations

import re
from typing import List, Optional

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

# DNS1035: must start with a LETTER (k8s.io/apimachinery IsDNS1035Label)
_DNS1035_RE = re.compile(r"^[a-z]([-a-z0-9]*[a-z0-9])?$")
# DNS1123 label: may start with a digit
_DNS1123_RE = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")

# constant.go:372-385 — bounded so generated child names never truncate
MAX_RAYCLUSTER_NAME_LEN = 53
MAX_RAYSERVICE_NAME_LEN = MAX_RAYCLUSTER_NAME_LEN - 6  # 47
MAX_RAYJOB_NAME_LEN = MAX_RAYCLUSTER_NAME_LEN - 6      # 47
MAX_RAYCRONJOB_NAME_LEN = MAX_RAYJOB_NAME_LEN - 11     # 36

COLLECTOR_CONTAINER_NAME = "history-collector"


def _features_enabled(gate: str) -> bool:
    from .. import features
    return features.enabled(gate)


def _env_value(env_list, name: str) -> Optional[str]:
    for e in env_list or []:
        if e.name == name:
            return e.value if e.value is not None else ""
    return None


def _env_exists(env_list, name: str) -> bool:
    return _env_value(env_list, name) is not None


def _head_container(spec: RayClusterSpec):
    containers = spec.head_group_spec.template.spec.containers
    return containers[0] if containers else None


def _autoscaler_v2(spec: RayClusterSpec) -> bool:
    if spec.autoscaler_options and spec.autoscaler_options.version == "v2":
        return True
    head = _head_container(spec)
    if head is not None:
        val = _env_value(head.env, C.RAY_ENABLE_AUTOSCALER_V2)
        if val is not None and str(val).lower() in ("1", "true"):
            return True
    return False


def _ray_version_tuple(ray_version: Optional[str]):
    if not ray_version:
        return None
    m = re.match(r"(\d+)\.(\d+)", ray_version)
    if m is None:
        raise ValueError(ray_version)
    return tuple(int(x) for x in m.groups())


# ---------------------------------------------------------------------------
# RayCluster
# ---------------------------------------------------------------------------

def validate_raycluster_status(cluster: RayCluster) -> List[str]:
    """validation.go:23-30 ValidateRayClusterStatus."""
    conds = {c.type: c.status for c in (cluster.status.conditions or [])}
    if conds.get("Suspending") == "True" and conds.get("Suspended") == "True":
        return ["invalid RayCluster state: Suspending and Suspended "
                "conditions should not be both true"]
    return []


def validate_raycluster_metadata(meta: ObjectMeta) -> List[str]:
    """validation.go:32-40 ValidateRayClusterMetadata."""
    errs = []
    name = meta.name or ""
    if len(name) > MAX_RAYCLUSTER_NAME_LEN:
        errs.append(f"RayCluster name should be no more than "
                    f"{MAX_RAYCLUSTER_NAME_LEN} characters")
    if name and not _DNS1035_RE.match(name):
        errs.append(f"RayCluster name '{name}' should be a valid DNS1035 label")
    return errs


def validate_raycluster_spec(cluster: RayCluster) -> List[str]:
    """validation.go:103-331 ValidateRayClusterSpec."""
    errs = []
    spec = cluster.spec
    annotations = cluster.metadata.annotations or {}
    if not spec.head_group_spec.template.spec.containers:
        errs.append("headGroupSpec should have at least one container")

    errs += _validate_group_resources_and_labels(
        "head", spec.head_group_spec.ray_start_params,
        spec.head_group_spec.resources, spec.head_group_spec.labels)

    autoscaling = bool(spec.enable_in_tree_autoscaling)
    group_names = set()
    for group in spec.worker_group_specs:
        name = group.group_name
        if not name:
            errs.append("workerGroupSpec groupName must not be empty")
        if name in group_names:
            errs.append(f"duplicate worker group name '{name}'")
        group_names.add(name)
        if not group.template.spec.containers:
            errs.append(f"worker group '{name}' should have at least one container")
        # validation.go:125-138 — negative / inverted replica bounds
        if group.min_replicas is not None and group.min_replicas < 0:
            errs.append(f"worker group {name} has negative minReplicas "
                        f"{group.min_replicas}")
        if group.max_replicas is not None and group.max_replicas < 0:
            errs.append(f"worker group {name} has negative maxReplicas "
                        f"{group.max_replicas}")
        if (group.min_replicas is not None and group.max_replicas is not None
                and group.min_replicas > group.max_replicas):
            errs.append(f"worker group {name} has minReplicas "
                        f"{group.min_replicas} greater than maxReplicas "
                        f"{group.max_replicas}")
        if group.replicas is not None and group.replicas < 0:
            errs.append(f"worker group '{name}': replicas must be >= 0")
        if group.num_of_hosts < 0:
            errs.append(f"worker group '{name}': numOfHosts must be >= 0")
        errs += _validate_group_resources_and_labels(
            name or "worker", group.ray_start_params, group.resources,
            group.labels)
        errs += _validate_worker_idle_timeout(group, spec)
        errs += _validate_worker_priority(group, spec)
        # validation.go:199-205 — suspend gated on RayJobDeletionPolicy
        if group.suspend and not _features_enabled("RayJobDeletionPolicy"):
            errs.append(f"worker group {name} can be suspended only when the "
                        "RayJobDeletionPolicy feature gate is enabled")
        # validation.go:216-222 — suspend unsupported with autoscaler
        if group.suspend and autoscaling:
            errs.append(f"worker group {name} cannot be suspended with "
                        "Autoscaler enabled")

    errs += _validate_gcs_ft(cluster)

    head = _head_container(spec)
    # validation.go:193-196 — redis username only via GcsFaultToleranceOptions
    if spec.head_group_spec.ray_start_params.get("redis-username") or (
            head is not None and _env_exists(head.env, C.REDIS_USERNAME)):
        errs.append("cannot set redis username in rayStartParams or "
                    "environment variables - use "
                    "GcsFaultToleranceOptions.RedisUsername instead")

    # validation.go:207-214 — RAY_enable_autoscaler_v2 without autoscaling
    if not autoscaling and head is not None:
        val = _env_value(head.env, C.RAY_ENABLE_AUTOSCALER_V2)
        if val is not None and str(val).lower() in ("1", "true"):
            errs.append(f"environment variable {C.RAY_ENABLE_AUTOSCALER_V2} "
                        f"cannot be set to '{val}' when "
                        "enableInTreeAutoscaling is false")

    if autoscaling:
        # validation.go:224-227 — version field vs env var
        if (spec.autoscaler_options and spec.autoscaler_options.version
                and head is not None
                and _env_exists(head.env, C.RAY_ENABLE_AUTOSCALER_V2)):
            errs.append("both .spec.autoscalerOptions.version and head Pod "
                        f"env var {C.RAY_ENABLE_AUTOSCALER_V2} are set, "
                        "please only use the former")
        # validation.go:229-241 — v2 requires restartPolicy Never/unset
        if _autoscaler_v2(spec):
            rp = spec.head_group_spec.template.spec.restart_policy
            if rp and rp != "Never":
                errs.append("restartPolicy for head Pod should be Never or "
                            "unset when using autoscaler V2")
            for group in spec.worker_group_specs:
                rp = group.template.spec.restart_policy
                if rp and rp != "Never":
                    errs.append(f"restartPolicy for worker group "
                                f"{group.group_name} should be Never or "
                                "unset when using autoscaler V2")

    # validation.go:245-249 — idleTimeoutSeconds non-negative
    ao = spec.autoscaler_options
    if ao is not None and ao.idle_timeout_seconds is not None \
            and ao.idle_timeout_seconds < 0:
        errs.append("autoscalerOptions.idleTimeoutSeconds must be "
                    f"non-negative, got {ao.idle_timeout_seconds}")
    # validation.go:256-262 — KUBERAY_GEN_AUTOSCALER_START_CMD is managed
    if ao is not None and _env_exists(ao.env, C.KUBERAY_GEN_AUTOSCALER_START_CMD):
        errs.append(f"autoscalerOptions.env must not contain "
                    f"{C.KUBERAY_GEN_AUTOSCALER_START_CMD}: it is managed by "
                    "the operator and injected automatically")

    errs += _validate_history_server(spec)
    errs += _validate_auth(spec)

    # validation.go:313-317 — networkPolicy behind its gate
    if spec.network_policy is not None and \
            not _features_enabled("RayClusterNetworkPolicy"):
        errs.append("spec.networkPolicy requires the RayClusterNetworkPolicy "
                    "feature gate to be enabled")
    errs += _validate_network_policy(spec)

    # validation.go:320-322 — tlsOptions behind its gate
    if spec.tls_options is not None and not _features_enabled("RayClusterMTLS"):
        errs.append("spec.tlsOptions requires the RayClusterMTLS feature "
                    "gate to be enabled")
    errs += _validate_tls(spec)
    errs += _validate_cluster_upgrade(cluster)
    return errs


def _validate_group_resources_and_labels(group_name: str, ray_start_params,
                                         resources, labels) -> List[str]:
    """validation.go:60-101 validateRayGroupResources/Labels."""
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


_LABEL_KEY_RE = re.compile(
    r"^([a-z0-9]([-a-z0-9.]*[a-z0-9])?/)?[A-Za-z0-9]([-A-Za-z0-9_.]*[A-Za-z0-9])?$")
_LABEL_VALUE_RE = re.compile(r"^([A-Za-z0-9]([-A-Za-z0-9_.]*[A-Za-z0-9])?)?$")


def _validate_gcs_ft(cluster: RayCluster) -> List[str]:
    """validation.go:153-192 + validateGcsFaultToleranceBackend :333-378.

    The reference's 'rocksdb' embedded backend is 'embedded' here (same
    semantics: GCS state on an operator-managed PVC instead of Redis).
    """
    errs: List[str] = []
    spec = cluster.spec
    opts = spec.gcs_fault_tolerance_options
    annotations = cluster.metadata.annotations or {}
    head = _head_container(spec)

    if annotations.get(C.RAY_FT_ENABLED_ANNOTATION_KEY) and opts is not None:
        errs.append(
            f"annotation {C.RAY_FT_ENABLED_ANNOTATION_KEY} and "
            "gcsFaultToleranceOptions are mutually exclusive")

    ft_enabled = opts is not None or bool(
        annotations.get(C.RAY_FT_ENABLED_ANNOTATION_KEY, "").lower() == "true")
    # validation.go:158-164 — RAY_REDIS_ADDRESS implies FT
    if not ft_enabled and head is not None and \
            _env_exists(head.env, C.RAY_REDIS_ADDRESS):
        errs.append(f"{C.RAY_REDIS_ADDRESS} is set which implicitly enables "
                    "GCS fault tolerance, but gcsFaultToleranceOptions is "
                    "not set")

    if opts is None:
        return errs

    # validation.go:168-186 — operator-owned fields must not be duplicated
    if spec.head_group_spec.ray_start_params.get("redis-password"):
        errs.append("cannot set `redis-password` in rayStartParams when "
                    "GcsFaultToleranceOptions is enabled - use "
                    "GcsFaultToleranceOptions.RedisPassword instead")
    if head is not None and _env_exists(head.env, C.REDIS_PASSWORD):
        errs.append("cannot set `REDIS_PASSWORD` env var in head Pod when "
                    "GcsFaultToleranceOptions is enabled - use "
                    "GcsFaultToleranceOptions.RedisPassword instead")
    if head is not None and _env_exists(head.env, C.RAY_REDIS_ADDRESS):
        errs.append("cannot set `RAY_REDIS_ADDRESS` env var in head Pod when "
                    "GcsFaultToleranceOptions is enabled - use "
                    "GcsFaultToleranceOptions.RedisAddress instead")
    if annotations.get(C.RAY_EXTERNAL_STORAGE_NS_ANNOTATION_KEY):
        errs.append("cannot set `ray.io/external-storage-namespace` "
                    "annotation when GcsFaultToleranceOptions is enabled - "
                    "use GcsFaultToleranceOptions.ExternalStorageNamespace "
                    "instead")

    backend = opts.backend or "redis"
    if backend in ("embedded", "rocksdb"):
        # validateGcsFaultToleranceBackend rocksdb branch (:337-369)
        if not _features_enabled("GCSFaultToleranceEmbeddedStorage"):
            errs.append("the embedded GCS fault tolerance backend requires "
                        "the GCSFaultToleranceEmbeddedStorage feature gate")
        if opts.redis_address:
            errs.append("cannot set GcsFaultToleranceOptions.RedisAddress "
                        "when backend is 'embedded'")
        if opts.redis_username is not None:
            errs.append("cannot set GcsFaultToleranceOptions.RedisUsername "
                        "when backend is 'embedded'")
        if opts.redis_password is not None:
            errs.append("cannot set GcsFaultToleranceOptions.RedisPassword "
                        "when backend is 'embedded'")
        if opts.external_storage_namespace:
            errs.append("cannot set GcsFaultToleranceOptions."
                        "ExternalStorageNamespace when backend is 'embedded'")
        storage = opts.storage
        if storage is not None and storage.claim_name:
            if storage.size is not None or storage.storage_class_name \
                    is not None or storage.access_modes:
                errs.append("GcsFaultToleranceOptions.Storage.ClaimName is "
                            "mutually exclusive with size, storageClassName, "
                            "and accessModes")
        if head is not None and (
                _env_exists(head.env, C.RAY_GCS_STORAGE)
                or _env_exists(head.env, C.RAY_GCS_STORAGE_PATH)):
            errs.append(f"cannot set `{C.RAY_GCS_STORAGE}` or "
                        f"`{C.RAY_GCS_STORAGE_PATH}` env var in head Pod "
                        "when the embedded GCS FT backend is used - these "
                        "are managed by the operator")
        if head is not None:
            for mount in head.volume_mounts or []:
                if mount.mount_path == C.GCS_STORAGE_MOUNT_PATH or \
                        mount.name == C.GCS_STORAGE_VOLUME_NAME:
                    errs.append("cannot set a volume mount named "
                                f"'{C.GCS_STORAGE_VOLUME_NAME}' or mounted "
                                f"at {C.GCS_STORAGE_MOUNT_PATH} in the head "
                                "container when the embedded GCS FT backend "
                                "is used - it is managed by the operator")
        for volume in spec.head_group_spec.template.spec.volumes or []:
            vname = volume.get("name") if isinstance(volume, dict) else \
                getattr(volume, "name", None)
            if vname == C.GCS_STORAGE_VOLUME_NAME:
                errs.append("cannot set a volume named "
                            f"'{C.GCS_STORAGE_VOLUME_NAME}' in the head Pod "
                            "when the embedded GCS FT backend is used - it "
                            "is managed by the operator")
    else:  # redis (default)
        if opts.storage is not None:
            errs.append("cannot set GcsFaultToleranceOptions.Storage when "
                        "backend is 'redis' - it only applies to the "
                        "embedded backend")
        if not opts.redis_address:
            errs.append("redis GCS backend requires redisAddress")
    return errs


def _validate_history_server(spec: RayClusterSpec) -> List[str]:
    """validation.go:264-287 + validateCollectorOptions :1029-1111."""
    errs: List[str] = []
    hso = spec.history_server_options
    if hso is None:
        return errs
    if not _features_enabled("RayClusterHistoryServer"):
        errs.append("RayClusterHistoryServer feature gate is not enabled")
    opts = hso.collector_options
    if opts is None:
        errs.append("historyServerOptions.collectorOptions must be set")
        return errs
    if not opts.image:
        errs.append("historyServerOptions.collectorOptions.image must be set")
    env = {e.name: e for e in (opts.env or [])}
    for name in sorted(_COLLECTOR_MANAGED_ENV & set(env)):
        errs.append(
            f"historyServerOptions.collectorOptions.env must not contain "
            f"{name}: it is injected by the operator")
    backend = env.get("STORAGE_BACKEND")
    if backend is None or not backend.value:
        errs.append("STORAGE_BACKEND environment variable must be set with a "
                    "literal string value in "
                    "historyServerOptions.collectorOptions.env")
    else:
        needed = _COLLECTOR_BACKEND_REQUIRES.get(backend.value.lower())
        if needed and not (env.get(needed) and
                           (env[needed].value or env[needed].value_from)):
            errs.append(
                f"{needed} env must be set when STORAGE_BACKEND is "
                f"{backend.value}")
    # :278-287 — reserved collector container name
    for container in spec.head_group_spec.template.spec.containers or []:
        if container.name == COLLECTOR_CONTAINER_NAME:
            errs.append(f"head pod template must not define a container "
                        f"named '{COLLECTOR_CONTAINER_NAME}' when history "
                        "server collector options are enabled")
    for group in spec.worker_group_specs:
        for container in group.template.spec.containers or []:
            if container.name == COLLECTOR_CONTAINER_NAME:
                errs.append(f"worker group {group.group_name} pod template "
                            "must not define a container named "
                            f"'{COLLECTOR_CONTAINER_NAME}' when history "
                            "server collector options are enabled")
    return errs


_COLLECTOR_MANAGED_ENV = {"POD_IP", "RAY_ROLE", "OWNER_KIND", "OWNER_NAME",
                          C.RAY_CLUSTER_NAMESPACE, "EVENTS_PORT"}
_COLLECTOR_BACKEND_REQUIRES = {
    "s3": "S3_REGION", "gcs": "GCS_BUCKET", "azure": "AZURE_STORAGE_ACCOUNT",
    "oss": "OSS_ENDPOINT"}


def _validate_auth(spec: RayClusterSpec) -> List[str]:
    """validation.go:289-311 — token auth needs Ray >= 2.52, K8s token auth
    >= 2.55 and no secretName; K8s token auth requires mode 'token'."""
    errs: List[str] = []
    ao = spec.auth_options
    if ao is None:
        return errs
    if ao.mode not in (None, "token", "disabled"):
        errs.append(f"authOptions.mode must be 'token' or 'disabled', "
                    f"got '{ao.mode}'")
    token_mode = ao.mode == "token"
    k8s_auth = bool(ao.enable_k8s_token_auth)
    if token_mode:
        if not spec.ray_version:
            errs.append("authOptions.mode is 'token' but rayVersion was not "
                        "specified. Ray version 2.52.0 or later is required")
            return errs
        try:
            version = _ray_version_tuple(spec.ray_version)
        except ValueError:
            errs.append("authOptions.mode is 'token' but rayVersion format "
                        f"is invalid: {spec.ray_version}")
            return errs
        if version < (2, 52):
            errs.append("authOptions.mode is 'token' but minimum Ray version "
                        f"is 2.52.0, got {spec.ray_version}")
        if k8s_auth:
            if version < (2, 55):
                errs.append("authOptions.enableK8sTokenAuth is enabled but "
                            "minimum Ray version is 2.55.0, got "
                            f"{spec.ray_version}")
            if ao.secret_name:
                errs.append("authOptions.enableK8sTokenAuth is enabled and "
                            "authOptions.secretName is also set")
    elif k8s_auth:
        errs.append("authOptions.enableK8sTokenAuth is enabled but "
                    "authOptions.mode not set to 'token'")
    return errs


def _validate_network_policy(spec: RayClusterSpec) -> List[str]:
    """validation.go:380-444 validateNetworkPolicy."""
    errs: List[str] = []
    np = spec.network_policy
    if np is None:
        return errs
    mode = np.mode or "DenyAll"
    if mode not in ("DenyAll", "DenyAllIngress", "DenyAllEgress"):
        errs.append(f"networkPolicy.mode invalid: '{np.mode}'")

    def check_rules(rules, where):
        if rules is None:
            return
        if mode == "DenyAllEgress" and rules.ingress_rules:
            errs.append(f"networkPolicy.{where}.ingressRules cannot be set "
                        f"when mode is '{mode}' (ingress is not restricted)")
        if mode == "DenyAllIngress" and rules.egress_rules:
            errs.append(f"networkPolicy.{where}.egressRules cannot be set "
                        f"when mode is '{mode}' (egress is not restricted)")

    check_rules(np.head, "head")
    check_rules(np.worker, "worker")
    # :420-429 — group names embedded in NetworkPolicy names need DNS1123
    group_names = set()
    for group in spec.worker_group_specs:
        if not _DNS1123_RE.match(group.group_name or ""):
            errs.append(f"worker group name '{group.group_name}' must be a "
                        "valid DNS1123 label when networkPolicy is enabled")
        group_names.add(group.group_name)
    for wg in np.worker_groups or []:
        check_rules(wg, f"workerGroups['{wg.group_name}']")
        if wg.group_name not in group_names:
            errs.append(f"networkPolicy.workerGroups['{wg.group_name}'] does "
                        "not match any group name in workerGroupSpecs")
    return errs


def _validate_tls(spec: RayClusterSpec) -> List[str]:
    """validation.go:446-535 validateTLSOptions."""
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
    # :488-506 — autoscalerOptions env/mounts conflict with managed TLS
    ao = spec.autoscaler_options
    if ao is not None:
        for env in ao.env or []:
            if env.name in forbidden:
                errs.append(f"cannot set {env.name} env in "
                            "autoscalerOptions.env when tlsOptions is "
                            "enabled — the operator manages TLS")
        for vm in ao.volume_mounts or []:
            if vm.name == C.RAY_TLS_VOLUME_NAME or \
                    vm.mount_path == C.RAY_TLS_CERT_MOUNT_PATH:
                errs.append("cannot use a volume mount named "
                            f"'{C.RAY_TLS_VOLUME_NAME}' or at "
                            f"{C.RAY_TLS_CERT_MOUNT_PATH} in "
                            "autoscalerOptions.volumeMounts when tlsOptions "
                            "is enabled")
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


def _validate_worker_idle_timeout(group, spec: RayClusterSpec) -> List[str]:
    """validation.go:1006-1027 validateWorkerGroupIdleTimeout."""
    errs: List[str] = []
    its = group.idle_timeout_seconds
    if its is None:
        return errs
    name = group.group_name or "worker"
    if its < 0:
        errs.append(f"worker group {name}: idleTimeoutSeconds must be "
                    f"non-negative, got {its}")
    if not _autoscaler_v2(spec):
        errs.append(
            f"worker group {name}: idleTimeoutSeconds is set, but autoscaler "
            "v2 is not enabled (set .spec.autoscalerOptions.version: v2)")
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
        parts = _ray_version_tuple(spec.ray_version)
    except ValueError:
        errs.append(f"worker group {name}: priority is set, but rayVersion "
                    f"format is invalid: {spec.ray_version}")
        return errs
    if parts < (2, 56):
        errs.append(f"worker group {name}: priority requires Ray >= 2.56.0, "
                    f"got {spec.ray_version}")
    if not _autoscaler_v2(spec):
        errs.append(f"worker group {name}: priority is only supported "
                    "with autoscaler v2 enabled")
    return errs


# ---------------------------------------------------------------------------
# RayJob (validation.go:527-658, :752-967)
# ---------------------------------------------------------------------------

def validate_rayjob_status(rayjob: RayJob) -> List[str]:
    """validation.go:527-532 ValidateRayJobStatus."""
    if rayjob.status.job_deployment_status == JobDeploymentStatus.WAITING \
            and rayjob.spec.submission_mode != JobSubmissionMode.INTERACTIVE:
        return ["JobDeploymentStatus cannot be `Waiting` when SubmissionMode "
                "is not InteractiveMode"]
    return []


def validate_rayjob_metadata(meta: ObjectMeta) -> List[str]:
    """validation.go:534-543 ValidateRayJobMetadata."""
    errs = []
    name = meta.name or ""
    if len(name) > MAX_RAYJOB_NAME_LEN:
        errs.append(f"RayJob name should be no more than "
                    f"{MAX_RAYJOB_NAME_LEN} characters")
    if name and not _DNS1035_RE.match(name):
        errs.append(f"RayJob name '{name}' should be a valid DNS1035 label")
    return errs


def validate_rayjob_spec(rayjob: RayJob) -> List[str]:
    """validation.go:545-658 ValidateRayJobSpec."""
    errs = []
    spec = rayjob.spec
    selector_mode = bool(spec.cluster_selector)
    if spec.submission_mode not in JobSubmissionMode.ALL:
        errs.append(f"invalid submissionMode '{spec.submission_mode}'")
    # :549-551 — suspend only with shutdownAfterJobFinishes
    if spec.suspend and not spec.shutdown_after_job_finishes:
        errs.append("a RayJob with shutdownAfterJobFinishes set to false is "
                    "not allowed to be suspended")
    if spec.ttl_seconds_after_finished < 0:
        errs.append("ttlSecondsAfterFinished must be >= 0")
    errs += _validate_deletion_configuration(rayjob)
    # :561-563 — suspend unsupported with clusterSelector
    if spec.suspend and selector_mode:
        errs.append("the ClusterSelector mode doesn't support the suspend "
                    "operation")
    if spec.ray_cluster_spec is None and not selector_mode:
        errs.append("one of rayClusterSpec or clusterSelector must be set")
    if spec.ray_cluster_spec is not None and selector_mode:
        errs.append("rayClusterSpec and clusterSelector are mutually exclusive")
    if selector_mode:
        # :568-571 — the ray.io/cluster key must name a cluster when present.
        # (Extension: arbitrary label selectors without the key are allowed
        # and re-resolved each reconcile — ops/rayjob.py.)
        if C.RAY_CLUSTER_LABEL_KEY in spec.cluster_selector and \
                not spec.cluster_selector[C.RAY_CLUSTER_LABEL_KEY]:
            errs.append("cluster name in ClusterSelector should not be empty")
        if spec.submission_mode == JobSubmissionMode.SIDECAR:
            errs.append("ClusterSelector is not supported in SidecarMode")
        # :575-577 — backoffLimit incompatible with clusterSelector
        if spec.backoff_limit is not None and spec.backoff_limit > 0:
            errs.append("BackoffLimit is incompatible with ClusterSelector mode")
        if spec.shutdown_after_job_finishes:
            errs.append("shutdownAfterJobFinishes is invalid with "
                        "clusterSelector (the job does not own the cluster)")
    if spec.submission_mode == JobSubmissionMode.INTERACTIVE:
        if spec.entrypoint:
            errs.append("entrypoint must be empty in InteractiveMode")
        # :587-589 — retries disallowed in InteractiveMode
        if spec.backoff_limit is not None and spec.backoff_limit > 0:
            errs.append("BackoffLimit is incompatible with InteractiveMode")
    elif not spec.entrypoint:
        errs.append("entrypoint is required unless submissionMode is "
                    "InteractiveMode")
    if spec.submission_mode == JobSubmissionMode.SIDECAR:
        # :591-603
        if spec.submitter_pod_template is not None:
            errs.append("SidecarMode doesn't support SubmitterPodTemplate")
        if spec.submitter_config is not None:
            errs.append("SidecarMode doesn't support SubmitterConfig")
        if spec.ray_cluster_spec is not None:
            rp = spec.ray_cluster_spec.head_group_spec.template.spec \
                .restart_policy
            if rp and rp != "Never":
                errs.append("restartPolicy for head Pod should be Never or "
                            "unset when using SidecarMode")
    if spec.submitter_pod_template is not None and spec.submission_mode not \
            in (JobSubmissionMode.K8S_JOB,):
        errs.append("submitterPodTemplate only applies to K8sJobMode")
    if spec.ray_cluster_spec is not None:
        # :606-608 — K8s token auth unsupported for RayJob
        ao = spec.ray_cluster_spec.auth_options
        if ao is not None and ao.enable_k8s_token_auth:
            errs.append("K8s token auth mode is currently not supported for "
                        "RayJob")
        sub = RayCluster(metadata=rayjob.metadata, spec=spec.ray_cluster_spec)
        errs += validate_raycluster_spec(sub)
    # :617-620 — runtimeEnvYAML must parse as YAML
    if spec.runtime_env_yaml:
        import yaml
        try:
            yaml.safe_load(spec.runtime_env_yaml)
        except yaml.YAMLError as e:
            errs.append(f"runtimeEnvYAML is not valid YAML: {e}")
    if spec.active_deadline_seconds is not None and \
            spec.active_deadline_seconds <= 0:
        errs.append("activeDeadlineSeconds must be a positive integer")
    # :624-626
    if spec.pre_running_deadline_seconds is not None and \
            spec.pre_running_deadline_seconds <= 0:
        errs.append("preRunningDeadlineSeconds must be a positive integer")
    if spec.backoff_limit is not None and spec.backoff_limit < 0:
        errs.append("backoffLimit must be >= 0")
    return errs


def _validate_deletion_configuration(rayjob: RayJob) -> List[str]:
    """validation.go:752-793 validateDeletionConfiguration."""
    errs: List[str] = []
    spec = rayjob.spec
    if not spec.shutdown_after_job_finishes and \
            spec.ttl_seconds_after_finished > 0:
        errs.append("a RayJob with shutdownAfterJobFinishes set to false "
                    "cannot have ttlSecondsAfterFinished")
    ds = spec.deletion_strategy
    if ds is None:
        return errs
    # :764-766 — feature gate guards any strategy use
    if not _features_enabled("RayJobDeletionPolicy"):
        errs.append("RayJobDeletionPolicy feature gate must be enabled to "
                    "use DeletionStrategy")
    legacy = ds.on_success is not None or ds.on_failure is not None
    rules = bool(ds.deletion_rules)
    if rules and spec.shutdown_after_job_finishes:
        errs.append("spec.shutdownAfterJobFinishes and "
                    "spec.deletionStrategy.deletionRules are mutually "
                    "exclusive")
    if rules and legacy:
        errs.append("Cannot use both legacy onSuccess/onFailure fields and "
                    "deletionRules simultaneously")
    if legacy:
        errs += _validate_legacy_deletion_policies(rayjob)
    elif rules:
        errs += _validate_deletion_rules(rayjob)
    else:
        errs.append("DeletionStrategy requires either BOTH onSuccess and "
                    "onFailure, OR the deletionRules field (cannot be empty)")
    return errs


_DELETION_ORDER = [DeletionPolicyType.DELETE_WORKERS,
                   DeletionPolicyType.DELETE_CLUSTER,
                   DeletionPolicyType.DELETE_SELF]


def _validate_deletion_rules(rayjob: RayJob) -> List[str]:
    """validation.go:795-900 validateDeletionRules + condition + TTL order."""
    errs: List[str] = []
    spec = rayjob.spec
    selector_mode = bool(spec.cluster_selector)
    autoscaling = bool(spec.ray_cluster_spec
                       and spec.ray_cluster_spec.enable_in_tree_autoscaling)
    by_job_status = {}
    by_deploy_status = {}
    for i, rule in enumerate(spec.deletion_strategy.deletion_rules or []):
        cond = rule.condition
        # validateDeletionCondition :869-887
        has_js = cond.job_status is not None
        has_jds = cond.job_deployment_status is not None
        if has_js and has_jds:
            errs.append(f"deletionRules[{i}]: cannot set both JobStatus and "
                        "JobDeploymentStatus at the same time")
            continue
        if not has_js and not has_jds:
            errs.append(f"deletionRules[{i}]: exactly one of JobStatus and "
                        "JobDeploymentStatus must be set")
            continue
        if cond.ttl_seconds < 0:
            errs.append(f"deletionRules[{i}]: TTLSeconds must be non-negative")
            continue
        if rule.policy not in DeletionPolicyType.ALL:
            errs.append(f"deletionRules[{i}]: invalid policy '{rule.policy}'")
            continue
        if has_js and cond.job_status not in JobStatus.TERMINAL:
            errs.append(f"deletionRules[{i}]: jobStatus "
                        f"'{cond.job_status}' is not terminal")
            continue
        if has_jds and cond.job_deployment_status not in (
                JobDeploymentStatus.COMPLETE, JobDeploymentStatus.FAILED):
            errs.append(f"deletionRules[{i}]: jobDeploymentStatus "
                        f"'{cond.job_deployment_status}' is not terminal")
            continue
        # contextual checks :831-840
        if selector_mode and rule.policy in (
                DeletionPolicyType.DELETE_CLUSTER,
                DeletionPolicyType.DELETE_WORKERS):
            errs.append(f"deletionRules[{i}]: DeletionPolicyType "
                        f"'{rule.policy}' not supported when ClusterSelector "
                        "is set")
            continue
        if autoscaling and rule.policy == DeletionPolicyType.DELETE_WORKERS:
            errs.append(f"deletionRules[{i}]: DeletionPolicyType "
                        "'DeleteWorkers' not supported with autoscaling "
                        "enabled")
            continue
        # uniqueness :842-866
        bucket = by_job_status.setdefault(cond.job_status, {}) if has_js \
            else by_deploy_status.setdefault(cond.job_deployment_status, {})
        key_desc = (f"JobStatus '{cond.job_status}'" if has_js
                    else f"JobDeploymentStatus '{cond.job_deployment_status}'")
        if rule.policy in bucket:
            errs.append(f"deletionRules[{i}]: duplicate rule for "
                        f"DeletionPolicyType '{rule.policy}' and {key_desc}")
            continue
        bucket[rule.policy] = cond.ttl_seconds
    # TTL consistency :889-923 — Workers <= Cluster <= Self
    for kind, buckets in (("JobStatus", by_job_status),
                          ("JobDeploymentStatus", by_deploy_status)):
        for value, policy_ttls in buckets.items():
            prev_policy = prev_ttl = None
            for policy in _DELETION_ORDER:
                if policy not in policy_ttls:
                    continue
                ttl = policy_ttls[policy]
                if prev_ttl is not None and ttl < prev_ttl:
                    errs.append(
                        f"for {kind} '{value}': {policy} TTL ({ttl}) must be "
                        f">= {prev_policy} TTL ({prev_ttl})")
                prev_policy, prev_ttl = policy, ttl
    return errs


def _validate_legacy_deletion_policies(rayjob: RayJob) -> List[str]:
    """validation.go:926-967 validateLegacyDeletionPolicies."""
    errs: List[str] = []
    ds = rayjob.spec.deletion_strategy
    selector_mode = bool(rayjob.spec.cluster_selector)
    if ds.on_success is None or ds.on_failure is None:
        errs.append("both DeletionStrategy.OnSuccess and "
                    "DeletionStrategy.OnFailure must be set when using the "
                    "legacy deletion policy fields")
        return errs
    for where, block in (("onSuccess", ds.on_success),
                         ("onFailure", ds.on_failure)):
        if block.policy is None:
            errs.append(f"the DeletionPolicyType field of "
                        f"DeletionStrategy.{where} cannot be unset")
            return errs
        if block.policy not in DeletionPolicyType.ALL:
            errs.append(f"deletionStrategy: invalid policy '{block.policy}'")
            return errs
    if selector_mode:
        for where, block in (("success", ds.on_success),
                             ("failure", ds.on_failure)):
            if block.policy in (DeletionPolicyType.DELETE_CLUSTER,
                                DeletionPolicyType.DELETE_WORKERS):
                errs.append(f"the ClusterSelector mode doesn't support "
                            f"DeletionStrategy={block.policy} on {where}")
    autoscaling = bool(rayjob.spec.ray_cluster_spec and
                       rayjob.spec.ray_cluster_spec.enable_in_tree_autoscaling)
    if autoscaling and DeletionPolicyType.DELETE_WORKERS in (
            ds.on_success.policy, ds.on_failure.policy):
        errs.append("DeletionStrategy=DeleteWorkers currently does not "
                    "support RayCluster with autoscaling enabled")
    if rayjob.spec.shutdown_after_job_finishes and \
            DeletionPolicyType.DELETE_NONE in (ds.on_success.policy,
                                               ds.on_failure.policy):
        errs.append("shutdownAfterJobFinishes is set to 'true' while "
                    "deletion policy is 'DeleteNone'")
    return errs


# ---------------------------------------------------------------------------
# RayService (validation.go:660-750)
# ---------------------------------------------------------------------------

def validate_rayservice_metadata(meta: ObjectMeta) -> List[str]:
    """validation.go:628-643 ValidateRayServiceMetadata."""
    errs = []
    name = meta.name or ""
    if len(name) > MAX_RAYSERVICE_NAME_LEN:
        errs.append(f"RayService name should be no more than "
                    f"{MAX_RAYSERVICE_NAME_LEN} characters")
    if name and not _DNS1035_RE.match(name):
        errs.append(f"RayService name '{name}' should be a valid DNS1035 label")
    errs += _validate_initializing_timeout(meta.annotations)
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
    """validation.go:680-716 + ValidateClusterUpgradeOptions :718-750."""
    errs = []
    spec = rayservice.spec
    errs += _validate_initializing_timeout(rayservice.metadata.annotations)
    # :681-683 — K8s token auth unsupported for RayService
    ao = spec.ray_cluster_spec.auth_options
    if ao is not None and ao.enable_k8s_token_auth:
        errs.append("K8s token auth mode is currently not supported for "
                    "RayService")
    sub = RayCluster(metadata=rayservice.metadata, spec=spec.ray_cluster_spec)
    errs += validate_raycluster_spec(sub)
    # :689-691 — headService name is operator-owned
    head_svc = spec.ray_cluster_spec.head_group_spec.head_service
    if head_svc is not None and head_svc.metadata.name:
        errs.append("spec.rayClusterConfig.headGroupSpec.headService."
                    "metadata.name should not be set")
    us = spec.upgrade_strategy
    if us is not None and us.type not in (
            None, RayServiceUpgradeType.NEW_CLUSTER,
            RayServiceUpgradeType.NEW_CLUSTER_WITH_INCREMENTAL_UPGRADE,
            RayServiceUpgradeType.NONE):
        errs.append(f"invalid upgradeStrategy.type '{us.type}'")
    # :702-706
    if spec.ray_cluster_deletion_delay_seconds is not None and \
            spec.ray_cluster_deletion_delay_seconds < 0:
        errs.append("rayClusterDeletionDelaySeconds should be a non-negative "
                    f"integer, got {spec.ray_cluster_deletion_delay_seconds}")
    if us is not None and us.type == \
            RayServiceUpgradeType.NEW_CLUSTER_WITH_INCREMENTAL_UPGRADE:
        errs += _validate_cluster_upgrade_options(rayservice)
    if spec.serve_config_v2:
        import yaml
        try:
            data = yaml.safe_load(spec.serve_config_v2)
            if not isinstance(data, dict) or "applications" not in data:
                errs.append("serveConfigV2 must be a YAML map with an "
                            "'applications' list")
        except yaml.YAMLError as e:
            errs.append(f"serveConfigV2 is not valid YAML: {e}")
    return errs


def _validate_cluster_upgrade_options(rayservice: RayService) -> List[str]:
    """validation.go:718-750 ValidateClusterUpgradeOptions."""
    errs: List[str] = []
    spec = rayservice.spec
    # :719-721 — incremental upgrade requires the autoscaler
    if not spec.ray_cluster_spec.enable_in_tree_autoscaling:
        errs.append("Ray Autoscaler is required for "
                    "NewClusterWithIncrementalUpgrade")
    opts = spec.upgrade_strategy.cluster_upgrade_options
    if opts is None:
        errs.append("ClusterUpgradeOptions are required for "
                    "NewClusterWithIncrementalUpgrade")
        return errs
    surge = opts.max_surge_percent if opts.max_surge_percent is not None \
        else 100
    if surge < 0 or surge > 100:
        errs.append("maxSurgePercent must be between 0 and 100")
    if opts.step_size_percent is None or not \
            (0 <= opts.step_size_percent <= 100):
        errs.append("stepSizePercent must be between 0 and 100")
    elif opts.step_size_percent > surge:
        errs.append("stepSizePercent must be less than or equal to "
                    "maxSurgePercent")
    if opts.interval_seconds is None or opts.interval_seconds <= 0:
        errs.append("intervalSeconds must be greater than 0")
    if not opts.gateway_class_name:
        errs.append("gatewayClassName is required for "
                    "NewClusterWithIncrementalUpgrade")
    return errs


# ---------------------------------------------------------------------------
# RayCronJob (validation.go:969-1004)
# ---------------------------------------------------------------------------

def validate_raycronjob_spec(cronjob: RayCronJob) -> List[str]:
    errs = []
    from .cron import parse_cron
    name = cronjob.metadata.name or ""
    # :971-973 — bounded so the deterministic child RayJob name stays valid
    if len(name) > MAX_RAYCRONJOB_NAME_LEN:
        errs.append(f"RayCronJob name should be no more than "
                    f"{MAX_RAYCRONJOB_NAME_LEN} characters")
    if not cronjob.spec.schedule:
        errs.append("schedule is required")
    elif "TZ" in cronjob.spec.schedule:
        # :976-978 — TZ/CRON_TZ prefixes are rejected; use timeZone
        errs.append("cannot use TZ or CRON_TZ in schedule, use timeZone "
                    "field instead")
    else:
        try:
            parse_cron(cronjob.spec.schedule)
        except ValueError as e:
            errs.append(f"invalid schedule: {e}")
    # :984-991 — empty-string timeZone is invalid (omit the field instead)
    if cronjob.spec.time_zone is not None:
        if cronjob.spec.time_zone == "":
            errs.append("timeZone must not be empty string, omit the field "
                        "to use the operator's local time zone")
        else:
            try:
                from zoneinfo import ZoneInfo
                ZoneInfo(cronjob.spec.time_zone)
            except Exception:
                errs.append(f"invalid timeZone {cronjob.spec.time_zone!r}")
    job = RayJob(metadata=cronjob.metadata, spec=cronjob.spec.job_template)
    errs += validate_rayjob_spec(job)
    return errs
