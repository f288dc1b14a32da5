"""Pod construction — where ALL the MI355X specificity lives.

Behavioral parity with the reference pod builder
(ray-operator/controllers/ray/common/pod.go: DefaultHeadPodTemplate :214,
DefaultWorkerPodTemplate :595, BuildPod :832, probes :721-829, env battery
:1207-1343, ray-start defaulting :1345-1430, GPU injection :1432-1479) with
these MI355X-native differences:

* ``amd.com/gpu`` is the only GPU resource key → ``--num-gpus``; the
  NVIDIA/MIG/TPU/neuron paths of the reference do not exist here,
* GPU pods get RCCL-over-xGMI env (NCCL_* names — RCCL reads them),
  ``HSA_ENABLE_IPC_MODE_LEGACY=0`` (dmabuf IPC), and an
  ``accelerator_type:AMD-Instinct-MI355X`` Ray custom resource,
* optional explicit ``/dev/kfd`` + ``/dev/dri`` hostPath mounts for nodes
  without the AMD device plugin,
* GPU worker readiness probes additionally run an on-device health gate
  (rocm-smi + gfx950 MFMA smoke kernel via ``python -m kuberay_amd.gpu.probe``)
  — the reference has no GPU-level probe at all,
* ``/dev/shm`` sizing defaults to the pod memory limit (MI355X hosts are
  large; plasma wants real shm).
"""
from __future__ import annotations

import copy
import json
import os
from typing import Dict, List, Optional

from ..kube.objects import (
    Container,
    ContainerPort,
    EnvVar,
    EnvVarSource,
    ExecAction,
    HTTPGetAction,
    Pod,
    PodTemplateSpec,
    Probe,
    ResourceRequirements,
    Volume,
    VolumeMount,
)
from ..models.raycluster import RayCluster, RayNodeType, WorkerGroupSpec
from ..utils import constants as C
from ..utils import names
from ..utils.resources import container_gpu_count, find_container_port

SHARED_MEMORY_MOUNT_PATH = "/dev/shm"
RAY_LOG_VOLUME_NAME = "ray-logs"
RAY_LOG_MOUNT_PATH = "/tmp/ray"
AUTOSCALER_CONTAINER_NAME = "autoscaler"
COLLECTOR_CONTAINER_NAME = "history-collector"
PLASMA_DIRECTORY_PARAM_KEY = "plasma-directory"
BASE_WGET_HEALTH_COMMAND = "wget --tries 1 -T %d -q -O- http://localhost:%d/%s | grep success"


# ---------------------------------------------------------------------------
# feature helpers (spec interrogation; util.go:751-800 analog)
# ---------------------------------------------------------------------------

def is_autoscaling_enabled(spec) -> bool:
    return bool(spec.enable_in_tree_autoscaling)


def is_autoscaling_v2_enabled(spec) -> bool:
    opts = spec.autoscaler_options
    return bool(opts and opts.version == "v2")


def is_gcs_fault_tolerance_enabled(cluster: RayCluster) -> bool:
    annotations = cluster.metadata.annotations or {}
    if annotations.get(C.RAY_FT_ENABLED_ANNOTATION_KEY, "").lower() == "true":
        return True
    return cluster.spec.gcs_fault_tolerance_options is not None


def is_auth_enabled(spec) -> bool:
    ao = spec.auth_options
    if ao is None:
        return False
    if ao.mode == "disabled":
        return False
    return bool(ao.enable_k8s_token_auth or ao.mode == "token" or ao.secret_name)


def is_tls_enabled(spec) -> bool:
    return bool(spec.tls_options and spec.tls_options.enabled)


def _env_flag(name: str, default: bool = True) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v.lower() == "true"


# ---------------------------------------------------------------------------
# labels & annotations
# ---------------------------------------------------------------------------

def label_pod(node_type: str, cluster_name: str, group_name: str,
              extra: Optional[Dict[str, str]] = None) -> Dict[str, str]:
    """pod.go labelPod analog: operator-owned labels win over user labels."""
    labels = dict(extra or {})
    labels.update({
        C.RAY_NODE_LABEL_KEY: "yes",
        C.RAY_CLUSTER_LABEL_KEY: names.check_label(cluster_name),
        C.RAY_NODE_TYPE_LABEL_KEY: node_type,
        C.RAY_NODE_GROUP_LABEL_KEY: names.check_label(group_name),
        C.RAY_ID_LABEL_KEY: names.check_label(names.identifier(cluster_name, node_type)),
        C.KUBERNETES_APPLICATION_NAME_LABEL_KEY: C.APPLICATION_NAME,
        C.KUBERNETES_CREATED_BY_LABEL_KEY: C.COMPONENT_NAME,
    })
    return labels


def _init_template_annotations(cluster: RayCluster, template: PodTemplateSpec) -> None:
    ann = template.metadata.ensure_annotations()
    if is_gcs_fault_tolerance_enabled(cluster):
        ann[C.RAY_FT_ENABLED_ANNOTATION_KEY] = "true"
        opts = cluster.spec.gcs_fault_tolerance_options
        ns = None
        if opts and opts.external_storage_namespace:
            ns = opts.external_storage_namespace
        elif (cluster.metadata.annotations or {}).get(C.RAY_EXTERNAL_STORAGE_NS_ANNOTATION_KEY):
            ns = cluster.metadata.annotations[C.RAY_EXTERNAL_STORAGE_NS_ANNOTATION_KEY]
        ann[C.RAY_EXTERNAL_STORAGE_NS_ANNOTATION_KEY] = ns or cluster.metadata.uid or ""
    else:
        ann[C.RAY_FT_ENABLED_ANNOTATION_KEY] = "false"


# ---------------------------------------------------------------------------
# ray start command
# ---------------------------------------------------------------------------

def set_missing_ray_start_params(params: Dict[str, str], node_type: str,
                                 head_port: str, fqdn_ray_ip: str) -> Dict[str, str]:
    """pod.go:1345 setMissingRayStartParams."""
    params = dict(params)
    if node_type == RayNodeType.WORKER and "address" not in params:
        params["address"] = f"{fqdn_ray_ip}:{head_port}"
    if node_type == RayNodeType.HEAD and "dashboard-host" not in params:
        params["dashboard-host"] = "0.0.0.0"
    if "metrics-export-port" not in params:
        params["metrics-export-port"] = str(C.DEFAULT_METRICS_PORT)
    params["block"] = "true"
    if "dashboard-agent-listen-port" not in params:
        params["dashboard-agent-listen-port"] = str(C.DEFAULT_DASHBOARD_AGENT_LISTEN_PORT)
    return params


def _int_quantity(value) -> int:
    from ..utils.quantity import parse_quantity
    import math
    return int(math.ceil(parse_quantity(value)))


def generate_ray_start_command(node_type: str, params: Dict[str, str],
                               resources: Optional[ResourceRequirements]) -> str:
    """pod.go:1392 generateRayStartCommand — amd.com/gpu only.

    Besides ``--num-gpus``, MI355X workers advertise the Ray custom resource
    ``accelerator_type:AMD-Instinct-MI355X`` so Ray applications can target
    the accelerator generation (Ray's accelerator_type convention).
    """
    params = dict(params)
    limits = (resources.limits if resources else None) or {}
    requests = (resources.requests if resources else None) or {}
    if "num-cpus" not in params:
        cpu = limits.get("cpu") or requests.get("cpu")
        if cpu is not None:
            params["num-cpus"] = str(_int_quantity(cpu))
    if "memory" not in params:
        mem = limits.get("memory")
        if mem is not None:
            params["memory"] = str(_int_quantity(mem))
    gpu = limits.get(C.AMD_GPU_RESOURCE_NAME) or requests.get(C.AMD_GPU_RESOURCE_NAME)
    ngpu = _int_quantity(gpu) if gpu is not None else 0
    if ngpu and "num-gpus" not in params:
        params["num-gpus"] = str(ngpu)
    if ngpu:
        resources_map = {}
        if "resources" in params:
            try:
                resources_map = json.loads(params["resources"].strip("'\""))
            except (ValueError, AttributeError):
                resources_map = {}
        key = f"accelerator_type:{C.RAY_ACCELERATOR_TYPE_AMD_MI355X}"
        if key not in resources_map:
            resources_map[key] = ngpu
            params["resources"] = "'" + json.dumps(resources_map, sort_keys=True) + "'"

    flags = " ".join(
        f"--{k}" if v == "true" and k in ("block", "head", "no-monitor", "include-dashboard")
        else f"--{k}={v}"
        for k, v in sorted(params.items())
    )
    prefix = "ray start --head " if node_type == RayNodeType.HEAD else "ray start "
    return prefix + flags


# ---------------------------------------------------------------------------
# MI355X GPU configuration
# ---------------------------------------------------------------------------

def configure_mi355x(template: PodTemplateSpec, *, mount_device_nodes: bool = False) -> int:
    """Inject ROCm/RCCL environment (and optionally /dev/kfd + /dev/dri
    hostPath mounts) into a pod that requests ``amd.com/gpu``.

    Returns the GPU count of the Ray container (0 → no-op). This replaces the
    reference's nvidia-oriented accelerator wiring (pod.go:40-49, :1432-1479)
    and adds what Ray-on-ROCm actually needs on an 8xMI355X node:

    * ``HSA_ENABLE_IPC_MODE_LEGACY=0`` — the host driver only supports
      dmabuf IPC; without it RCCL / CUDA-tensor sharing across processes
      fails with ``hipIpcGetMemHandle: invalid argument``,
    * RCCL env keeping collectives on the 7x~153 GB/s xGMI links (no IB,
      no net fallback, p2p+shm on),
    * NOTE: HIP_VISIBLE_DEVICES / ROCR_VISIBLE_DEVICES are *not* set — the
      AMD device plugin narrows the cgroup device set per pod; setting them
      here would fight the plugin's allocation.
    """
    ray_container = template.spec.containers[C.RAY_CONTAINER_INDEX]
    ngpu = container_gpu_count(ray_container)
    if ngpu == 0:
        return 0
    for name, value in C.RCCL_ENV_DEFAULTS.items():
        ray_container.set_env_if_absent(name, value)
    if mount_device_nodes:
        template.spec.add_volume_if_absent(Volume(
            name=C.DEV_KFD_VOLUME_NAME, host_path={"path": C.DEV_KFD_PATH, "type": "CharDevice"}))
        template.spec.add_volume_if_absent(Volume(
            name=C.DEV_DRI_VOLUME_NAME, host_path={"path": C.DEV_DRI_PATH, "type": "Directory"}))
        ray_container.add_volume_mount_if_absent(
            VolumeMount(name=C.DEV_KFD_VOLUME_NAME, mount_path=C.DEV_KFD_PATH))
        ray_container.add_volume_mount_if_absent(
            VolumeMount(name=C.DEV_DRI_VOLUME_NAME, mount_path=C.DEV_DRI_PATH))
        if ray_container.security_context is None:
            from ..kube.objects import SecurityContext
            ray_container.security_context = SecurityContext()
        caps = ray_container.security_context.capabilities or {}
        add = set(caps.get("add") or [])
        add.add("SYS_PTRACE")  # ROCm debug/IPC needs it
        caps["add"] = sorted(add)
        ray_container.security_context.capabilities = caps
    return ngpu


# ---------------------------------------------------------------------------
# probes
# ---------------------------------------------------------------------------

def _supports_unified_health_check(ray_version: Optional[str]) -> bool:
    """Ray >= 2.53 exposes one HTTP health endpoint (pod.go:721-731)."""
    if not ray_version:
        return False
    try:
        parts = [int(p) for p in ray_version.split(".")[:2]]
        return tuple(parts) >= (2, 53)
    except ValueError:
        return False


def init_liveness_and_readiness_probe(
    ray_container: Container, node_type: str, creator_crd_type: Optional[str],
    ray_start_params: Dict[str, str], ray_version: Optional[str],
    gpu_probe: bool = False,
) -> None:
    """pod.go:732 initLivenessAndReadinessProbe + MI355X GPU health gate.

    ``gpu_probe=True`` (GPU worker) appends an on-device check to the
    readiness exec: rocm-smi liveness + the gfx950 MFMA smoke kernel via
    ``python -m kuberay_amd.gpu.probe`` — catching wedged GPUs / HBM ECC
    fallout that HTTP probes can't see.
    """
    def get_port(key: str, default: int) -> int:
        try:
            return int(ray_start_params.get(key, default))
        except (TypeError, ValueError):
            return default

    agent_port = get_port("dashboard-agent-listen-port", C.DEFAULT_DASHBOARD_AGENT_LISTEN_PORT)
    dashboard_port = get_port("dashboard-port", C.DEFAULT_DASHBOARD_PORT)

    unified = _supports_unified_health_check(ray_version)
    http_action = HTTPGetAction(path="/api/local_raylet_healthz", port=agent_port)

    raylet_cmd = BASE_WGET_HEALTH_COMMAND % (
        C.DEFAULT_READINESS_PROBE_TIMEOUT_SECONDS, agent_port, C.RAY_AGENT_RAYLET_HEALTH_PATH)
    gcs_cmd = BASE_WGET_HEALTH_COMMAND % (
        C.DEFAULT_READINESS_PROBE_FAILURE_THRESHOLD, dashboard_port, C.RAY_DASHBOARD_GCS_HEALTH_PATH)

    commands = [raylet_cmd, gcs_cmd] if node_type == RayNodeType.HEAD else [raylet_cmd]

    if ray_container.liveness_probe is None:
        timeout = (C.DEFAULT_HEAD_LIVENESS_PROBE_TIMEOUT_SECONDS
                   if node_type == RayNodeType.HEAD
                   else C.DEFAULT_LIVENESS_PROBE_TIMEOUT_SECONDS)
        probe = Probe(
            initial_delay_seconds=C.DEFAULT_LIVENESS_PROBE_INITIAL_DELAY_SECONDS,
            timeout_seconds=timeout,
            period_seconds=C.DEFAULT_LIVENESS_PROBE_PERIOD_SECONDS,
            success_threshold=C.DEFAULT_LIVENESS_PROBE_SUCCESS_THRESHOLD,
            failure_threshold=C.DEFAULT_LIVENESS_PROBE_FAILURE_THRESHOLD,
        )
        if unified:
            probe.http_get = http_action
        else:
            probe.exec_ = ExecAction(command=["bash", "-c", " && ".join(commands)])
        ray_container.liveness_probe = probe

    if ray_container.readiness_probe is None:
        timeout = (C.DEFAULT_HEAD_READINESS_PROBE_TIMEOUT_SECONDS
                   if node_type == RayNodeType.HEAD
                   else C.DEFAULT_READINESS_PROBE_TIMEOUT_SECONDS)
        probe = Probe(
            initial_delay_seconds=C.DEFAULT_READINESS_PROBE_INITIAL_DELAY_SECONDS,
            timeout_seconds=timeout,
            period_seconds=C.DEFAULT_READINESS_PROBE_PERIOD_SECONDS,
            success_threshold=C.DEFAULT_READINESS_PROBE_SUCCESS_THRESHOLD,
            failure_threshold=C.DEFAULT_READINESS_PROBE_FAILURE_THRESHOLD,
        )
        ready_commands = list(commands)
        if creator_crd_type == C.KIND_RAYSERVICE and node_type == RayNodeType.WORKER:
            # serve-traffic workers: proxy healthz folded into readiness
            probe.failure_threshold = C.SERVE_READINESS_PROBE_FAILURE_THRESHOLD
            serve_port = find_container_port(ray_container, C.SERVING_PORT_NAME, C.DEFAULT_SERVING_PORT)
            ready_commands.append(BASE_WGET_HEALTH_COMMAND % (
                C.DEFAULT_READINESS_PROBE_INITIAL_DELAY_SECONDS, serve_port,
                C.RAY_SERVE_PROXY_HEALTH_PATH))
        if gpu_probe:
            # MI355X on-device gate; exec (not HTTP) so we always use exec here.
            ready_commands.append("python -m kuberay_amd.gpu.probe --quick")
            probe.exec_ = ExecAction(command=["bash", "-c", " && ".join(ready_commands)])
        elif unified and len(ready_commands) == len(commands):
            probe.http_get = http_action
        else:
            probe.exec_ = ExecAction(command=["bash", "-c", " && ".join(ready_commands)])
        ray_container.readiness_probe = probe


# ---------------------------------------------------------------------------
# shared volumes
# ---------------------------------------------------------------------------

def _add_empty_dir(container: Container, template_spec, name: str, path: str,
                   medium: Optional[str] = None, size_limit: Optional[str] = None) -> None:
    empty = {}
    if medium:
        empty["medium"] = medium
    if size_limit:
        empty["sizeLimit"] = size_limit
    template_spec.add_volume_if_absent(Volume(name=name, empty_dir=empty))
    container.add_volume_mount_if_absent(VolumeMount(name=name, mount_path=path))


# ---------------------------------------------------------------------------
# GCS fault tolerance, auth, TLS
# ---------------------------------------------------------------------------

def configure_gcs_fault_tolerance(template: PodTemplateSpec, cluster: RayCluster,
                                  node_type: str) -> None:
    """pod.go:117-211 configureGCSFaultTolerance (redis + embedded rocksdb)."""
    if not is_gcs_fault_tolerance_enabled(cluster):
        return
    container = template.spec.containers[C.RAY_CONTAINER_INDEX]
    opts = cluster.spec.gcs_fault_tolerance_options
    if node_type == RayNodeType.HEAD:
        ns = (opts.external_storage_namespace if opts and opts.external_storage_namespace
              else (cluster.metadata.annotations or {}).get(
                  C.RAY_EXTERNAL_STORAGE_NS_ANNOTATION_KEY) or cluster.metadata.uid or "")
        container.set_env_if_absent(C.RAY_EXTERNAL_STORAGE_NS, ns)
        if opts and opts.backend == "embedded":
            container.set_env(C.RAY_GCS_STORAGE, C.GCS_STORAGE_ROCKSDB_VALUE)
            storage = opts.storage
            sub_path = (storage.sub_path if storage and storage.sub_path else "")
            path = C.GCS_STORAGE_MOUNT_PATH + ("/" + sub_path if sub_path else "")
            container.set_env(C.RAY_GCS_STORAGE_PATH, path)
            claim = (storage.claim_name if storage and storage.claim_name
                     else names.gcs_pvc_name(cluster.metadata.name))
            template.spec.add_volume_if_absent(Volume(
                name=C.GCS_STORAGE_VOLUME_NAME,
                persistent_volume_claim={"claimName": claim}))
            container.add_volume_mount_if_absent(VolumeMount(
                name=C.GCS_STORAGE_VOLUME_NAME, mount_path=C.GCS_STORAGE_MOUNT_PATH))
        elif opts:
            if opts.redis_address:
                container.set_env_if_absent(C.RAY_REDIS_ADDRESS, opts.redis_address)
            if opts.redis_username:
                _set_credential_env(container, C.REDIS_USERNAME, opts.redis_username)
            if opts.redis_password:
                _set_credential_env(container, C.REDIS_PASSWORD, opts.redis_password)
    else:
        # workers tolerate a GCS restart window (pod.go:94-102)
        container.set_env_if_absent(
            C.RAY_GCS_RPC_SERVER_RECONNECT_TIMEOUT_S,
            C.DEFAULT_WORKER_RAY_GCS_RECONNECT_TIMEOUT_S)


def _set_credential_env(container: Container, name: str, cred) -> None:
    if cred.value_from is not None:
        if container.get_env(name) is None:
            container.env = container.env or []
            container.env.append(EnvVar(name=name, value_from=EnvVarSource.from_dict(cred.value_from)))
    elif cred.value:
        container.set_env_if_absent(name, cred.value)


def configure_token_auth(cluster_name: str, template: PodTemplateSpec, auth_options) -> None:
    """pod.go:345-426 — mount the auth token secret + env."""
    secret_name = (auth_options.secret_name if auth_options and auth_options.secret_name
                   else names.auth_secret_name(cluster_name))
    template.spec.add_volume_if_absent(Volume(
        name=C.RAY_TOKEN_VOLUME_NAME,
        secret={"secretName": secret_name}))
    for container in template.spec.containers:
        container.add_volume_mount_if_absent(VolumeMount(
            name=C.RAY_TOKEN_VOLUME_NAME, mount_path=C.RAY_TOKEN_MOUNT_PATH, read_only=True))
        container.set_env_if_absent(C.RAY_AUTH_MODE_ENV_VAR, "token")
        if container.get_env(C.RAY_AUTH_TOKEN_ENV_VAR) is None:
            container.env = container.env or []
            container.env.append(EnvVar(
                name=C.RAY_AUTH_TOKEN_ENV_VAR,
                value_from=EnvVarSource(secret_key_ref={
                    "name": secret_name, "key": C.RAY_AUTH_TOKEN_SECRET_KEY})))


def configure_tls(template: PodTemplateSpec, cluster: RayCluster, node_type: str) -> None:
    """pod.go:431-542 — mount cert secret + TLS env (cert-manager flow)."""
    if not is_tls_enabled(cluster.spec):
        return
    prefix = "ray-head-secret" if node_type == RayNodeType.HEAD else "ray-worker-secret"
    secret_name = f"{prefix}-{cluster.metadata.name}"
    template.spec.add_volume_if_absent(Volume(
        name=C.RAY_TLS_VOLUME_NAME, secret={"secretName": secret_name}))
    for container in template.spec.containers:
        container.add_volume_mount_if_absent(VolumeMount(
            name=C.RAY_TLS_VOLUME_NAME, mount_path=C.RAY_TLS_CERT_MOUNT_PATH, read_only=True))
        container.set_env_if_absent(C.RAY_USE_TLS, "1")
        container.set_env_if_absent(C.RAY_TLS_SERVER_CERT, f"{C.RAY_TLS_CERT_MOUNT_PATH}/tls.crt")
        container.set_env_if_absent(C.RAY_TLS_SERVER_KEY, f"{C.RAY_TLS_CERT_MOUNT_PATH}/tls.key")
        container.set_env_if_absent(C.RAY_TLS_CA_CERT, f"{C.RAY_TLS_CERT_MOUNT_PATH}/ca.crt")


# ---------------------------------------------------------------------------
# autoscaler sidecar
# ---------------------------------------------------------------------------

def build_autoscaler_container(autoscaler_image: str) -> Container:
    """pod.go:937 BuildAutoscalerContainer."""
    start_cmd = ("ray kuberay-autoscaler --cluster-name $(RAY_CLUSTER_NAME) "
                 "--cluster-namespace $(RAY_CLUSTER_NAMESPACE)")
    return Container(
        name=AUTOSCALER_CONTAINER_NAME,
        image=autoscaler_image,
        image_pull_policy="IfNotPresent",
        env=[
            EnvVar(name=C.RAY_CLUSTER_NAME, value_from=EnvVarSource(
                field_ref={"fieldPath": f"metadata.labels['{C.RAY_CLUSTER_LABEL_KEY}']"})),
            EnvVar(name=C.RAY_CLUSTER_NAMESPACE, value_from=EnvVarSource(
                field_ref={"fieldPath": "metadata.namespace"})),
            EnvVar(name="RAY_HEAD_POD_NAME", value_from=EnvVarSource(
                field_ref={"fieldPath": "metadata.name"})),
            EnvVar(name="KUBERAY_CRD_VER", value="v1"),
            EnvVar(name=C.KUBERAY_GEN_AUTOSCALER_START_CMD, value=start_cmd),
        ],
        command=["/bin/bash", "-lc", "--"],
        args=[start_cmd],
        resources=ResourceRequirements(
            limits={"cpu": "500m", "memory": "512Mi"},
            requests={"cpu": "500m", "memory": "512Mi"},
        ),
    )


def _merge_autoscaler_overrides(container: Container, options) -> None:
    if options is None:
        return
    if options.resources is not None:
        container.resources = options.resources
    if options.image:
        container.image = options.image
    if options.image_pull_policy:
        container.image_pull_policy = options.image_pull_policy
    if options.security_context is not None:
        container.security_context = options.security_context
    if options.env:
        for e in options.env:
            container.set_env_if_absent(e.name, e.value or "")
    if options.volume_mounts:
        for vm in options.volume_mounts:
            container.add_volume_mount_if_absent(vm)
    if options.command:
        container.command = options.command
    if options.args:
        container.args = options.args


# ---------------------------------------------------------------------------
# templates
# ---------------------------------------------------------------------------

def head_service_account_name(cluster: RayCluster) -> str:
    sa = cluster.spec.head_group_spec.template.spec.service_account_name
    return sa or cluster.metadata.name


def default_head_pod_template(cluster: RayCluster, head_spec, pod_name: str,
                              head_port: str) -> PodTemplateSpec:
    """pod.go:214 DefaultHeadPodTemplate."""
    template = head_spec.template.clone()
    if _env_flag(C.ENABLE_DETERMINISTIC_HEAD_POD_NAME, default=False):
        template.metadata.name = pod_name
        template.metadata.generate_name = None
    else:
        template.metadata.name = None
        template.metadata.generate_name = pod_name
    template.metadata.namespace = cluster.metadata.namespace or "default"

    params = dict(head_spec.ray_start_params)
    if head_spec.resources:
        _merge_params_resources(params, head_spec.resources)
    if head_spec.labels:
        _merge_params_labels(params, head_spec.labels)

    merged = dict(template.metadata.labels or {})
    merged.update(head_spec.labels or {})
    template.metadata.labels = label_pod(
        RayNodeType.HEAD, cluster.metadata.name, "headgroup", merged)

    if is_autoscaling_enabled(cluster.spec):
        params["no-monitor"] = "true"
        template.spec.service_account_name = names.check_name(
            head_service_account_name(cluster))
        ray_image = template.spec.containers[C.RAY_CONTAINER_INDEX].image
        autoscaler = build_autoscaler_container(ray_image)
        if is_auth_enabled(cluster.spec):
            configure_token_auth_container(cluster.metadata.name, autoscaler,
                                           cluster.spec.auth_options)
        _merge_autoscaler_overrides(autoscaler, cluster.spec.autoscaler_options)
        template.spec.containers.append(autoscaler)
        if is_autoscaling_v2_enabled(cluster.spec):
            template.spec.containers[C.RAY_CONTAINER_INDEX].set_env_if_absent(
                C.RAY_ENABLE_AUTOSCALER_V2, "true")
            template.spec.restart_policy = "Never"

    head_spec.ray_start_params = params  # defaulting is visible to callers (reference semantics)

    # history-server collector sidecar (pod.go:304-315; gated)
    from .. import features
    if (features.enabled("RayClusterHistoryServer")
            and cluster.spec.history_server_options is not None
            and cluster.spec.history_server_options.collector_options is not None):
        from ..historyserver.collector import build_collector_container
        collector = build_collector_container(
            cluster.spec.history_server_options.collector_options,
            RayNodeType.HEAD, cluster.metadata.name,
            cluster.metadata.namespace or "default",
            names.fqdn_service_name(cluster, cluster.metadata.namespace or "default"))
        if collector.image is None:
            collector.image = template.spec.containers[C.RAY_CONTAINER_INDEX].image
        template.spec.containers.append(collector)
        ray_c = template.spec.containers[C.RAY_CONTAINER_INDEX]
        ray_c.set_env_if_absent("RAY_enable_ray_event", "true")
        ray_c.set_env_if_absent(
            "RAY_enable_core_worker_ray_event_to_aggregator", "true")

    configure_gcs_fault_tolerance(template, cluster, RayNodeType.HEAD)
    _ensure_metrics_port(template)
    if is_auth_enabled(cluster.spec):
        configure_token_auth(cluster.metadata.name, template, cluster.spec.auth_options)
    configure_tls(template, cluster, RayNodeType.HEAD)
    _init_template_annotations(cluster, template)
    configure_mi355x(template)
    return template


def configure_token_auth_container(cluster_name: str, container: Container, auth_options) -> None:
    secret_name = (auth_options.secret_name if auth_options and auth_options.secret_name
                   else names.auth_secret_name(cluster_name))
    container.set_env_if_absent(C.RAY_AUTH_MODE_ENV_VAR, "token")
    if container.get_env(C.RAY_AUTH_TOKEN_ENV_VAR) is None:
        container.env = container.env or []
        container.env.append(EnvVar(
            name=C.RAY_AUTH_TOKEN_ENV_VAR,
            value_from=EnvVarSource(secret_key_ref={
                "name": secret_name, "key": C.RAY_AUTH_TOKEN_SECRET_KEY})))


def default_worker_pod_template(cluster: RayCluster, worker_spec: WorkerGroupSpec,
                                pod_name: str, fqdn_ray_ip: str, head_port: str,
                                replica_grp_name: str = "", replica_index: int = 0,
                                num_host_index: int = 0) -> PodTemplateSpec:
    """pod.go:595 DefaultWorkerPodTemplate (incl. wait-gcs-ready init container
    :606-658 and multi-host labels :673-682)."""
    template = worker_spec.template.clone()
    template.metadata.name = None
    template.metadata.generate_name = pod_name
    template.metadata.namespace = cluster.metadata.namespace or "default"

    if _env_flag(C.ENABLE_INIT_CONTAINER_INJECTION, default=True):
        ray_container = template.spec.containers[C.RAY_CONTAINER_INDEX]
        wait_script = (
            "SECONDS=0; "
            "until ray health-check --address "
            f"{fqdn_ray_ip}:{head_port} > /dev/null 2>&1; do "
            'echo "$SECONDS seconds elapsed: Waiting for GCS to be ready."; sleep 5; done; '
            'echo "GCS is ready."'
        )
        init = Container(
            name="wait-gcs-ready",
            image=ray_container.image,
            image_pull_policy=ray_container.image_pull_policy,
            command=["/bin/bash", "-lc", "--"],
            args=[wait_script],
            security_context=copy.deepcopy(ray_container.security_context),
            env=copy.deepcopy(ray_container.env),
            volume_mounts=copy.deepcopy(ray_container.volume_mounts),
            resources=ResourceRequirements(
                limits={"cpu": "200m", "memory": "256Mi"},
                requests={"cpu": "200m", "memory": "256Mi"},
            ),
        )
        template.spec.init_containers = (template.spec.init_containers or []) + [init]

    params = dict(worker_spec.ray_start_params)
    if worker_spec.resources:
        _merge_params_resources(params, worker_spec.resources)
    if worker_spec.labels:
        _merge_params_labels(params, worker_spec.labels)
    worker_spec.ray_start_params = params

    merged = dict(template.metadata.labels or {})
    merged.update(worker_spec.labels or {})
    template.metadata.labels = label_pod(
        RayNodeType.WORKER, cluster.metadata.name, worker_spec.group_name, merged)
    template.metadata.labels[C.RAY_WORKER_REPLICA_INDEX_KEY] = str(replica_index)
    if worker_spec.num_of_hosts > 1:
        template.metadata.labels[C.RAY_WORKER_REPLICA_NAME_KEY] = replica_grp_name
        template.metadata.labels[C.RAY_HOST_INDEX_KEY] = str(num_host_index)

    configure_gcs_fault_tolerance(template, cluster, RayNodeType.WORKER)
    _ensure_metrics_port(template)
    if is_autoscaling_enabled(cluster.spec) and is_autoscaling_v2_enabled(cluster.spec):
        template.spec.restart_policy = "Never"
    if is_auth_enabled(cluster.spec):
        configure_token_auth(cluster.metadata.name, template, cluster.spec.auth_options)
    configure_tls(template, cluster, RayNodeType.WORKER)
    _init_template_annotations(cluster, template)
    configure_mi355x(template)
    return template


def _ensure_metrics_port(template: PodTemplateSpec) -> None:
    container = template.spec.containers[C.RAY_CONTAINER_INDEX]
    if find_container_port(container, C.METRICS_PORT_NAME, -1) == -1:
        container.ports = (container.ports or []) + [
            ContainerPort(name=C.METRICS_PORT_NAME, container_port=C.DEFAULT_METRICS_PORT)]


def _merge_params_resources(params: Dict[str, str], resources: Dict[str, str]) -> None:
    """Top-level group Resources field → ray start --resources JSON."""
    existing = {}
    if "resources" in params:
        try:
            existing = json.loads(params["resources"].strip("'\""))
        except (ValueError, AttributeError):
            existing = {}
    for k, v in resources.items():
        if k not in existing:
            try:
                existing[k] = float(v)
            except ValueError:
                continue
    if existing:
        params["resources"] = "'" + json.dumps(existing, sort_keys=True) + "'"


def _merge_params_labels(params: Dict[str, str], labels: Dict[str, str]) -> None:
    existing = {}
    if "labels" in params:
        try:
            existing = json.loads(params["labels"].strip("'\""))
        except (ValueError, AttributeError):
            existing = {}
    for k, v in labels.items():
        existing.setdefault(k, v)
    if existing:
        params["labels"] = "'" + json.dumps(existing, sort_keys=True) + "'"


# ---------------------------------------------------------------------------
# BuildPod
# ---------------------------------------------------------------------------

def build_pod(
    template: PodTemplateSpec,
    node_type: str,
    ray_start_params: Dict[str, str],
    head_port: str,
    enable_ray_autoscaler: bool,
    creator_crd_type: Optional[str],
    fqdn_ray_ip: str,
    default_container_envs: Optional[List[EnvVar]] = None,
    ray_version: Optional[str] = None,
) -> Pod:
    """pod.go:832 BuildPod — template → concrete Pod."""
    template = template.clone()
    labels = template.metadata.ensure_labels()
    if creator_crd_type == C.KIND_RAYSERVICE:
        labels[C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY] = (
            C.ENABLE_RAY_CLUSTER_SERVING_SERVICE_FALSE
            if node_type == RayNodeType.HEAD
            else C.ENABLE_RAY_CLUSTER_SERVING_SERVICE_TRUE)

    pod = Pod(metadata=template.metadata, spec=template.spec)
    ray_container = pod.spec.containers[C.RAY_CONTAINER_INDEX]

    # /dev/shm for plasma, sized to the memory limit (MI355X hosts are big)
    params = set_missing_ray_start_params(ray_start_params, node_type, head_port, fqdn_ray_ip)
    if PLASMA_DIRECTORY_PARAM_KEY not in params:
        size = None
        if ray_container.resources and ray_container.resources.limits:
            size = ray_container.resources.limits.get("memory")
        _add_empty_dir(ray_container, pod.spec, C.SHARED_MEMORY_VOLUME_NAME,
                       SHARED_MEMORY_MOUNT_PATH, medium="Memory",
                       size_limit=str(size) if size else None)

    if node_type == RayNodeType.HEAD and enable_ray_autoscaler:
        for c in pod.spec.containers:
            if c.name == AUTOSCALER_CONTAINER_NAME:
                _add_empty_dir(ray_container, pod.spec, RAY_LOG_VOLUME_NAME, RAY_LOG_MOUNT_PATH)
                _add_empty_dir(c, pod.spec, RAY_LOG_VOLUME_NAME, RAY_LOG_MOUNT_PATH)

    # command synthesis
    user_cmd = " ".join(ray_container.command or [])
    if ray_container.args:
        user_cmd = (user_cmd + " " + " ".join(ray_container.args)).strip()
    ray_start_cmd = generate_ray_start_command(node_type, params, ray_container.resources)
    ulimit_cmd = f"ulimit -n ${{{C.RAY_START_ULIMIT_OPEN_FILES}:-65536}}"

    overwrite = (template.metadata.annotations or {}).get(
        C.RAY_OVERWRITE_CONTAINER_CMD_ANNOTATION_KEY, "").lower() == "true"
    if not overwrite and "ray start" not in user_cmd:
        generated = f"{ulimit_cmd}; {ray_start_cmd}"
        args = f"{user_cmd} && {generated}" if user_cmd else generated
        login_shell = _env_flag(C.ENABLE_LOGIN_SHELL, default=False)
        ray_container.command = ["/bin/bash", "-lc" if login_shell else "-c", "--"]
        ray_container.args = [args]

    for init in pod.spec.init_containers or []:
        init.set_env_if_absent(C.FQ_RAY_IP, fqdn_ray_ip)
        init.set_env_if_absent(C.RAY_IP, names.extract_ray_ip_from_fqdn(fqdn_ray_ip))

    _set_container_env_vars(pod, node_type, fqdn_ray_ip, head_port, ray_start_cmd,
                            creator_crd_type, default_container_envs or [])

    if _env_flag(C.ENABLE_PROBES_INJECTION, default=True):
        gpu_probe = (node_type == RayNodeType.WORKER
                     and container_gpu_count(ray_container) > 0)
        init_liveness_and_readiness_probe(
            ray_container, node_type, creator_crd_type, params, ray_version,
            gpu_probe=gpu_probe)
    return pod


def _set_container_env_vars(pod: Pod, node_type: str, fqdn_ray_ip: str, head_port: str,
                            ray_start_cmd: str, creator_crd_type: Optional[str],
                            default_envs: List[EnvVar]) -> None:
    """pod.go:1207 setContainerEnvVars."""
    container = pod.spec.containers[C.RAY_CONTAINER_INDEX]
    for e in default_envs:
        if container.get_env(e.name) is None:
            container.env = container.env or []
            container.env.append(e.clone())

    ip = C.LOCAL_HOST
    if node_type == RayNodeType.WORKER:
        ip = fqdn_ray_ip
        container.set_env_if_absent(C.FQ_RAY_IP, ip)
        container.set_env_if_absent(C.RAY_IP, names.extract_ray_ip_from_fqdn(ip))

    container.env = container.env or []
    container.env.append(EnvVar(name=C.RAY_CLUSTER_NAME, value_from=EnvVarSource(
        field_ref={"fieldPath": f"metadata.labels['{C.RAY_CLUSTER_LABEL_KEY}']"})))
    container.env.append(EnvVar(name=C.RAY_CLUSTER_NAMESPACE, value_from=EnvVarSource(
        field_ref={"fieldPath": "metadata.namespace"})))
    container.env.append(EnvVar(name=C.RAY_CLOUD_INSTANCE_ID, value_from=EnvVarSource(
        field_ref={"fieldPath": "metadata.name"})))
    container.env.append(EnvVar(name=C.RAY_NODE_TYPE_NAME, value_from=EnvVarSource(
        field_ref={"fieldPath": f"metadata.labels['{C.RAY_NODE_GROUP_LABEL_KEY}']"})))
    container.env.append(EnvVar(name=C.KUBERAY_GEN_RAY_START_CMD, value=ray_start_cmd))
    container.set_env_if_absent(C.RAY_PORT, head_port)

    if creator_crd_type == C.KIND_RAYSERVICE:
        container.set_env_if_absent(C.RAY_TIMEOUT_MS_TASK_WAIT_FOR_DEATH_INFO, "0")
        container.set_env_if_absent(C.RAY_GCS_SERVER_REQUEST_TIMEOUT_SECONDS, "5")
        container.set_env_if_absent(C.RAY_SERVE_KV_TIMEOUT_S, "5")

    container.set_env_if_absent(C.RAY_ADDRESS, f"{ip}:{head_port}")
    container.set_env_if_absent(C.RAY_USAGE_STATS_KUBERAY_IN_USE, "1")
    if node_type == RayNodeType.HEAD:
        from .. import KUBERAY_VERSION
        container.set_env_if_absent(
            C.RAY_USAGE_STATS_EXTRA_TAGS,
            f"kuberay_version={KUBERAY_VERSION};kuberay_crd={creator_crd_type or C.KIND_RAYCLUSTER}")
    container.set_env_if_absent(C.RAY_DASHBOARD_ENABLE_K8S_DISK_USAGE, "1")


def get_head_port(head_start_params: Dict[str, str]) -> str:
    return head_start_params.get("port", str(C.DEFAULT_GCS_SERVER_PORT))
