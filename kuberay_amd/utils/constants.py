"""API-surface constants of the ray.io/v1 contract, plus MI355X-native additions.

The label/annotation/port/env-var names mirror the reference operator's public
surface (ray-operator/controllers/ray/utils/constant.go) so that existing
sample YAMLs, clients and Ray images work verbatim against this operator.
Everything NVIDIA-specific in the reference is intentionally absent; the
MI355X/ROCm block at the bottom is new.
"""

# ---------------------------------------------------------------------------
# Group / version / kinds
# ---------------------------------------------------------------------------
GROUP = "ray.io"
VERSION = "v1"
API_VERSION = f"{GROUP}/{VERSION}"

KIND_RAYCLUSTER = "RayCluster"
KIND_RAYJOB = "RayJob"
KIND_RAYSERVICE = "RayService"
KIND_RAYCRONJOB = "RayCronJob"

# ---------------------------------------------------------------------------
# Labels (reference: utils/constant.go:17-56)
# ---------------------------------------------------------------------------
RAY_ORIGINATED_FROM_CR_NAME_LABEL_KEY = "ray.io/originated-from-cr-name"
RAY_ORIGINATED_FROM_CRD_LABEL_KEY = "ray.io/originated-from-crd"
RAY_CLUSTER_LABEL_KEY = "ray.io/cluster"
RAY_NODE_TYPE_LABEL_KEY = "ray.io/node-type"
RAY_NODE_GROUP_LABEL_KEY = "ray.io/group"
RAY_NODE_LABEL_KEY = "ray.io/is-ray-node"
RAY_ID_LABEL_KEY = "ray.io/identifier"
RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY = "ray.io/serve"
RAY_CLUSTER_HEADLESS_SERVICE_LABEL_KEY = "ray.io/headless-worker-svc"
HASH_WITHOUT_REPLICAS_AND_WORKERS_TO_DELETE_KEY = (
    "ray.io/hash-without-replicas-and-workers-to-delete"
)
UPGRADE_STRATEGY_RECREATE_HASH_KEY = "ray.io/upgrade-strategy-recreate-hash"
NUM_WORKER_GROUPS_KEY = "ray.io/num-worker-groups"
KUBERAY_VERSION_LABEL_KEY = "ray.io/kuberay-version"
RAY_CRONJOB_NAME_LABEL_KEY = "ray.io/cronjob-name"
RAY_JOB_SUBMISSION_MODE_LABEL_KEY = "ray.io/job-submission-mode"
RAY_WORKER_REPLICA_NAME_KEY = "ray.io/worker-group-replica-name"
RAY_WORKER_REPLICA_INDEX_KEY = "ray.io/worker-group-replica-index"
RAY_HOST_INDEX_KEY = "ray.io/replica-host-index"
RAY_PRIORITY_CLASS_NAME = "ray.io/priority-class-name"
RAY_GANG_SCHEDULING_ENABLED = "ray.io/gang-scheduling-enabled"
KUBERNETES_APPLICATION_NAME_LABEL_KEY = "app.kubernetes.io/name"
KUBERNETES_CREATED_BY_LABEL_KEY = "app.kubernetes.io/created-by"

# label values
HEAD_NODE = "head"
WORKER_NODE = "worker"
APPLICATION_NAME = "kuberay"
COMPONENT_NAME = "kuberay-operator"
ENABLE_RAY_CLUSTER_SERVING_SERVICE_TRUE = "true"
ENABLE_RAY_CLUSTER_SERVING_SERVICE_FALSE = "false"

# ---------------------------------------------------------------------------
# Annotations
# ---------------------------------------------------------------------------
RAY_FT_ENABLED_ANNOTATION_KEY = "ray.io/ft-enabled"
RAY_EXTERNAL_STORAGE_NS_ANNOTATION_KEY = "ray.io/external-storage-namespace"
RAY_CLUSTER_GCS_FT_DELETION_TIMEOUT_ANNOTATION = "ray.io/gcs-ft-deletion-timeout"
RAY_OVERWRITE_CONTAINER_CMD_ANNOTATION_KEY = "ray.io/overwrite-container-cmd"
RAY_SERVICE_INITIALIZING_TIMEOUT_ANNOTATION = "ray.io/initializing-timeout"
RAY_CRONJOB_TIMESTAMP_ANNOTATION_KEY = "ray.io/cronjob-scheduled-timestamp"
DISABLE_PROVISIONED_HEAD_RESTART_ANNOTATION_KEY = (
    "ray.io/disable-provisioned-head-restart"
)
ENABLE_SERVE_SERVICE_KEY = "ray.io/enable-serve-service"
ENABLE_SERVE_SERVICE_TRUE = "true"

# ---------------------------------------------------------------------------
# Finalizers
# ---------------------------------------------------------------------------
GCS_FT_REDIS_CLEANUP_FINALIZER = "ray.io/gcs-ft-redis-cleanup-finalizer"
RAYJOB_STOP_JOB_FINALIZER = "ray.io/rayjob-finalizer"
RAY_CLUSTER_SELECTOR_KEY = "ray.io/cluster"

# ---------------------------------------------------------------------------
# Default ports (reference: utils/constant.go:105-120)
# ---------------------------------------------------------------------------
DEFAULT_CLIENT_PORT = 10001
DEFAULT_GCS_SERVER_PORT = 6379
DEFAULT_DASHBOARD_PORT = 8265
DEFAULT_METRICS_PORT = 8080
DEFAULT_DASHBOARD_AGENT_LISTEN_PORT = 52365
DEFAULT_SERVING_PORT = 8000

CLIENT_PORT_NAME = "client"
GCS_SERVER_PORT_NAME = "gcs-server"
DASHBOARD_PORT_NAME = "dashboard"
METRICS_PORT_NAME = "metrics"
SERVING_PORT_NAME = "serve"
DASHBOARD_AGENT_PORT_NAME = "dashboard-agent"

DEFAULT_SERVICE_APP_PROTOCOL = "tcp"
HEADLESS_SERVICE_SUFFIX = "headless"
DASH = "-"

# ---------------------------------------------------------------------------
# Container env vars injected into Ray pods (reference: constant.go:138-200)
# ---------------------------------------------------------------------------
RAY_CLUSTER_NAME = "RAY_CLUSTER_NAME"
RAY_CLUSTER_NAMESPACE = "RAY_CLUSTER_NAMESPACE"
RAY_IP = "RAY_IP"
FQ_RAY_IP = "FQ_RAY_IP"
RAY_PORT = "RAY_PORT"
RAY_ADDRESS = "RAY_ADDRESS"
RAY_REDIS_ADDRESS = "RAY_REDIS_ADDRESS"
REDIS_PASSWORD = "REDIS_PASSWORD"
REDIS_USERNAME = "REDIS_USERNAME"
RAY_EXTERNAL_STORAGE_NS = "RAY_external_storage_namespace"
RAY_GCS_STORAGE = "RAY_gcs_storage"
RAY_GCS_STORAGE_PATH = "RAY_gcs_storage_path"
RAY_GCS_RPC_SERVER_RECONNECT_TIMEOUT_S = "RAY_gcs_rpc_server_reconnect_timeout_s"
RAY_TIMEOUT_MS_TASK_WAIT_FOR_DEATH_INFO = "RAY_timeout_ms_task_wait_for_death_info"
RAY_GCS_SERVER_REQUEST_TIMEOUT_SECONDS = "RAY_gcs_server_request_timeout_seconds"
RAY_SERVE_KV_TIMEOUT_S = "RAY_SERVE_KV_TIMEOUT_S"
RAY_USAGE_STATS_KUBERAY_IN_USE = "RAY_USAGE_STATS_KUBERAY_IN_USE"
RAY_USAGE_STATS_EXTRA_TAGS = "RAY_USAGE_STATS_EXTRA_TAGS"
KUBERAY_GEN_RAY_START_CMD = "KUBERAY_GEN_RAY_START_CMD"
KUBERAY_GEN_AUTOSCALER_START_CMD = "KUBERAY_GEN_AUTOSCALER_START_CMD"
RAY_START_ULIMIT_OPEN_FILES = "RAY_START_ULIMIT_OPEN_FILES"
RAY_DASHBOARD_ENABLE_K8S_DISK_USAGE = "RAY_DASHBOARD_ENABLE_K8S_DISK_USAGE"
RAY_CLOUD_INSTANCE_ID = "RAY_CLOUD_INSTANCE_ID"
RAY_NODE_TYPE_NAME = "RAY_NODE_TYPE_NAME"
RAY_ENABLE_AUTOSCALER_V2 = "RAY_enable_autoscaler_v2"
RAY_DASHBOARD_ADDRESS = "RAY_DASHBOARD_ADDRESS"
RAY_JOB_SUBMISSION_ID = "RAY_JOB_SUBMISSION_ID"
RAY_USE_TLS = "RAY_USE_TLS"
RAY_TLS_SERVER_CERT = "RAY_TLS_SERVER_CERT"
RAY_TLS_SERVER_KEY = "RAY_TLS_SERVER_KEY"
RAY_TLS_CA_CERT = "RAY_TLS_CA_CERT"
RAY_AUTH_MODE_ENV_VAR = "RAY_AUTH_MODE"
RAY_AUTH_TOKEN_ENV_VAR = "RAY_AUTH_TOKEN"
RAY_AUTH_TOKEN_SECRET_KEY = "auth_token"

DEFAULT_WORKER_RAY_GCS_RECONNECT_TIMEOUT_S = "600"
LOCAL_HOST = "127.0.0.1"

# TLS volume / mounts
RAY_TLS_VOLUME_NAME = "ray-tls"
RAY_TLS_CERT_MOUNT_PATH = "/etc/ray/tls"
RAY_TOKEN_VOLUME_NAME = "ray-token"
RAY_TOKEN_MOUNT_PATH = "/var/run/secrets/ray.io/serviceaccount"

# GCS embedded storage (RocksDB on PVC)
GCS_STORAGE_VOLUME_NAME = "gcs-storage"
GCS_STORAGE_MOUNT_PATH = "/data/gcs"
GCS_STORAGE_ROCKSDB_VALUE = "rocksdb"
GCS_STORAGE_PVC_SUFFIX = "-gcs-pvc"
GCS_STORAGE_DEFAULT_SIZE = "1Gi"

# ---------------------------------------------------------------------------
# Probe defaults (reference: constant.go:300-330)
# ---------------------------------------------------------------------------
DEFAULT_READINESS_PROBE_INITIAL_DELAY_SECONDS = 10
DEFAULT_READINESS_PROBE_TIMEOUT_SECONDS = 2
DEFAULT_HEAD_READINESS_PROBE_TIMEOUT_SECONDS = 5
DEFAULT_READINESS_PROBE_PERIOD_SECONDS = 5
DEFAULT_READINESS_PROBE_SUCCESS_THRESHOLD = 1
DEFAULT_READINESS_PROBE_FAILURE_THRESHOLD = 10
SERVE_READINESS_PROBE_FAILURE_THRESHOLD = 1
DEFAULT_LIVENESS_PROBE_INITIAL_DELAY_SECONDS = 30
DEFAULT_LIVENESS_PROBE_TIMEOUT_SECONDS = 2
DEFAULT_HEAD_LIVENESS_PROBE_TIMEOUT_SECONDS = 5
DEFAULT_LIVENESS_PROBE_PERIOD_SECONDS = 5
DEFAULT_LIVENESS_PROBE_SUCCESS_THRESHOLD = 1
DEFAULT_LIVENESS_PROBE_FAILURE_THRESHOLD = 120

RAY_AGENT_RAYLET_HEALTH_PATH = "api/local_raylet_healthz"
RAY_DASHBOARD_GCS_HEALTH_PATH = "api/gcs_healthz"
RAY_SERVE_PROXY_HEALTH_PATH = "-/healthz"

# ---------------------------------------------------------------------------
# Operator env-var feature flags (legacy toggles; reference: constant.go)
# ---------------------------------------------------------------------------
ENABLE_RANDOM_POD_DELETE = "ENABLE_RANDOM_POD_DELETE"
ENABLE_GCS_FT_REDIS_CLEANUP = "ENABLE_GCS_FT_REDIS_CLEANUP"
ENABLE_PROBES_INJECTION = "ENABLE_PROBES_INJECTION"
ENABLE_INIT_CONTAINER_INJECTION = "ENABLE_INIT_CONTAINER_INJECTION"
ENABLE_RAY_HEAD_CLUSTER_IP_SERVICE = "ENABLE_RAY_HEAD_CLUSTER_IP_SERVICE"
ENABLE_LOGIN_SHELL = "ENABLE_LOGIN_SHELL"
ENABLE_DETERMINISTIC_HEAD_POD_NAME = "ENABLE_DETERMINISTIC_HEAD_POD_NAME"
DELETE_RAYJOB_CR_AFTER_JOB_FINISHES = "DELETE_RAYJOB_CR_AFTER_JOB_FINISHES"
RAYCLUSTER_DEFAULT_REQUEUE_SECONDS_ENV = "RAYCLUSTER_DEFAULT_REQUEUE_SECONDS_ENV"
RAYCLUSTER_DEFAULT_REQUEUE_SECONDS = 300
RAYJOB_DEPLOYMENT_STATUS_TRANSITION_GRACE_PERIOD_SECONDS = (
    "RAYJOB_DEPLOYMENT_STATUS_TRANSITION_GRACE_PERIOD_SECONDS"
)
DEFAULT_RAYJOB_DEPLOYMENT_STATUS_TRANSITION_GRACE_PERIOD_SECONDS = 300
RAYJOB_STATUS_CHECK_TIMEOUT_SECONDS = "RAYJOB_STATUS_CHECK_TIMEOUT_SECONDS"
DEFAULT_RAYJOB_STATUS_CHECK_TIMEOUT_SECONDS = 300
RAYCLUSTER_GCS_FT_DELETION_TIMEOUT_DEFAULT = 300

# Ray container is always the first app container in a head/worker pod.
RAY_CONTAINER_INDEX = 0

# ---------------------------------------------------------------------------
# MI355X / ROCm-native surface (no NVIDIA analogs kept).
#
# Reference touchpoints replaced:
#   utils/resources.go:8-17  (GPU key detection)   -> AMD_GPU_RESOURCE_NAME only
#   common/pod.go:1432-1479  (num-gpus injection)  -> keyed on amd.com/gpu
#   common/pod.go:40-49      (custom accelerators) -> dropped
# ---------------------------------------------------------------------------
AMD_GPU_RESOURCE_NAME = "amd.com/gpu"
# node labels published by kuberay_amd.gpu.labeller, consumed by the
# XgmiGangScheduler for topology-aware gang placement
XGMI_ISLAND_NODE_LABEL = "amd.com/xgmi-island"
AMD_GPU_COUNT_LABEL = "amd.com/gpu.count"
XGMI_FULLY_CONNECTED_LABEL = "amd.com/xgmi-fully-connected"
XGMI_LARGEST_ISLAND_LABEL = "amd.com/xgmi-largest-island"

# Device nodes every ROCm container needs (AMD k8s device plugin mounts these
# automatically when amd.com/gpu is requested; we also support explicit
# hostPath injection for clusters without the device plugin).
DEV_KFD_PATH = "/dev/kfd"
DEV_DRI_PATH = "/dev/dri"
DEV_KFD_VOLUME_NAME = "dev-kfd"
DEV_DRI_VOLUME_NAME = "dev-dri"

# ROCm visibility / runtime env for Ray containers.
HIP_VISIBLE_DEVICES = "HIP_VISIBLE_DEVICES"
ROCR_VISIBLE_DEVICES = "ROCR_VISIBLE_DEVICES"
HSA_ENABLE_IPC_MODE_LEGACY = "HSA_ENABLE_IPC_MODE_LEGACY"

# RCCL-over-xGMI environment injected into every GPU worker (RCCL reads
# NCCL_*-named variables). On a single 8xMI355X node the fabric is 7 p2p
# xGMI links per GPU; there is no IB/RoCE plumbing to configure in v1.
RCCL_ENV_DEFAULTS = {
    # dmabuf IPC is the only mode the MI355X host driver supports.
    HSA_ENABLE_IPC_MODE_LEGACY: "0",
    # Never fall back to TCP sockets for intra-node traffic.
    "NCCL_NET_DISABLE": "0",
    "NCCL_IB_DISABLE": "1",
    # xGMI p2p is always preferable intra-node.
    "NCCL_P2P_DISABLE": "0",
    "NCCL_SHM_DISABLE": "0",
    # MSCCL++/RCCL tuning that benefits per-link-bound ring collectives on
    # 7x153GB/s xGMI: keep channels high enough to saturate all links.
    "NCCL_MIN_NCHANNELS": "28",
}

# Ray resource name under which Ray schedules AMD GPUs (Ray uses the generic
# "GPU" resource; accelerator type advertised via this custom resource).
RAY_ACCELERATOR_TYPE_AMD_MI355X = "AMD-Instinct-MI355X"

# rocm-smi based GPU health probing (replaces nothing in the reference —
# KubeRay has no GPU-level probe; BASELINE config #2 requires one).
ROCM_SMI_BIN = "/opt/rocm/bin/rocm-smi"
AMD_SMI_BIN = "/opt/rocm/bin/amd-smi"

# Default ROCm Ray image for samples / generation.
DEFAULT_RAY_ROCM_IMAGE = "rayproject/ray:2.46.0-py310-rocm624"

# /dev/shm object-store volume (plasma) — same mechanism as the reference
# (common/pod.go:855-861) but sized for MI355X hosts (large host RAM).
SHARED_MEMORY_VOLUME_NAME = "shared-mem"
