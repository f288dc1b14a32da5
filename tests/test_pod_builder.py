"""Unit tests for the pod builder (reference analog: common/pod_test.go).

MI355X-specific assertions: amd.com/gpu → --num-gpus, RCCL env injection,
GPU readiness gate, device mounts — all assertable on generated pod specs
without a GPU.
"""
import pytest

from kuberay_amd.common import pod as podlib
from kuberay_amd.common import service as servicelib
from kuberay_amd.models import RayCluster, RayNodeType
from kuberay_amd.testing import simple_raycluster
from kuberay_amd.utils import constants as C


def build_worker_pod(cluster, creator=None):
    group = cluster.spec.worker_group_specs[0]
    fqdn = "demo-head-svc.default.svc.cluster.local"
    template = podlib.default_worker_pod_template(cluster, group, "demo-worker-", fqdn, "6379")
    return podlib.build_pod(template, RayNodeType.WORKER, group.ray_start_params,
                            "6379", False, creator, fqdn,
                            ray_version=cluster.spec.ray_version)


def build_head_pod(cluster, creator=None):
    head = cluster.spec.head_group_spec
    template = podlib.default_head_pod_template(cluster, head, "demo-head-", "6379")
    return podlib.build_pod(template, RayNodeType.HEAD, head.ray_start_params, "6379",
                            podlib.is_autoscaling_enabled(cluster.spec), creator, "",
                            ray_version=cluster.spec.ray_version)


class TestRayStartCommand:
    def test_head_command(self):
        cluster = simple_raycluster("demo")
        pod = build_head_pod(cluster)
        args = pod.spec.containers[0].args[0]
        assert "ray start --head" in args
        assert "--block" in args
        assert "--dashboard-host=0.0.0.0" in args
        assert "--metrics-export-port=8080" in args
        assert "ulimit -n ${RAY_START_ULIMIT_OPEN_FILES:-65536}" in args

    def test_worker_address_points_to_head_fqdn(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        args = pod.spec.containers[0].args[0]
        assert "--address=demo-head-svc.default.svc.cluster.local:6379" in args
        assert "--head" not in args

    def test_amd_gpu_num_gpus_injection(self):
        cluster = simple_raycluster("demo", gpus_per_worker=4)
        pod = build_worker_pod(cluster)
        args = pod.spec.containers[0].args[0]
        assert "--num-gpus=4" in args
        assert C.RAY_ACCELERATOR_TYPE_AMD_MI355X in args

    def test_no_nvidia_resource_recognized(self):
        cluster = simple_raycluster("demo")
        cluster.spec.worker_group_specs[0].template.spec.containers[0].resources.limits[
            "nvidia.com/gpu"] = "8"
        pod = build_worker_pod(cluster)
        args = pod.spec.containers[0].args[0]
        assert "--num-gpus" not in args  # no dual-vendor dispatch

    def test_user_command_preserved_before_ray_start(self):
        cluster = simple_raycluster("demo")
        cluster.spec.worker_group_specs[0].template.spec.containers[0].command = ["echo", "hi"]
        pod = build_worker_pod(cluster)
        args = pod.spec.containers[0].args[0]
        assert args.startswith("echo hi && ")

    def test_overwrite_annotation_respected(self):
        cluster = simple_raycluster("demo")
        group = cluster.spec.worker_group_specs[0]
        group.template.metadata.annotations = {
            C.RAY_OVERWRITE_CONTAINER_CMD_ANNOTATION_KEY: "true"}
        group.template.spec.containers[0].command = ["mycmd"]
        pod = build_worker_pod(cluster)
        assert pod.spec.containers[0].command == ["mycmd"]
        # generated cmd still stored in env
        env = pod.spec.containers[0].get_env(C.KUBERAY_GEN_RAY_START_CMD)
        assert env is not None and "ray start" in env.value


class TestMI355XInjection:
    def test_rccl_env_injected_for_gpu_worker(self):
        cluster = simple_raycluster("demo", gpus_per_worker=1)
        pod = build_worker_pod(cluster)
        env = {e.name: e.value for e in pod.spec.containers[0].env}
        assert env[C.HSA_ENABLE_IPC_MODE_LEGACY] == "0"
        assert env["NCCL_IB_DISABLE"] == "1"
        assert env["NCCL_P2P_DISABLE"] == "0"

    def test_no_rccl_env_for_cpu_worker(self):
        cluster = simple_raycluster("demo", gpus_per_worker=0)
        pod = build_worker_pod(cluster)
        env_names = pod.spec.containers[0].env_names()
        assert "NCCL_IB_DISABLE" not in env_names

    def test_gpu_readiness_probe_includes_device_gate(self):
        cluster = simple_raycluster("demo", gpus_per_worker=1)
        pod = build_worker_pod(cluster)
        probe = pod.spec.containers[0].readiness_probe
        assert probe.exec_ is not None
        assert "kuberay_amd.gpu.probe" in probe.exec_.command[-1]

    def test_cpu_worker_probe_has_no_gpu_gate(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        probe = pod.spec.containers[0].readiness_probe
        cmd = probe.exec_.command[-1] if probe.exec_ else ""
        assert "kuberay_amd.gpu.probe" not in cmd

    def test_device_node_mounts_optional(self):
        cluster = simple_raycluster("demo", gpus_per_worker=1)
        group = cluster.spec.worker_group_specs[0]
        template = group.template.clone()
        n = podlib.configure_mi355x(template, mount_device_nodes=True)
        assert n == 1
        vols = {v.name for v in template.spec.volumes or []}
        assert C.DEV_KFD_VOLUME_NAME in vols and C.DEV_DRI_VOLUME_NAME in vols
        mounts = {m.mount_path for m in template.spec.containers[0].volume_mounts or []}
        assert C.DEV_KFD_PATH in mounts and C.DEV_DRI_PATH in mounts


class TestEnvBattery:
    def test_core_env_vars(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        c = pod.spec.containers[0]
        names = c.env_names()
        for expected in (C.RAY_CLUSTER_NAME, C.RAY_CLUSTER_NAMESPACE,
                         C.RAY_CLOUD_INSTANCE_ID, C.RAY_NODE_TYPE_NAME,
                         C.KUBERAY_GEN_RAY_START_CMD, C.RAY_PORT, C.RAY_ADDRESS,
                         C.FQ_RAY_IP, C.RAY_IP, C.RAY_USAGE_STATS_KUBERAY_IN_USE):
            assert expected in names, expected
        assert c.get_env(C.RAY_ADDRESS).value == \
            "demo-head-svc.default.svc.cluster.local:6379"

    def test_head_uses_localhost_address(self):
        cluster = simple_raycluster("demo")
        pod = build_head_pod(cluster)
        assert pod.spec.containers[0].get_env(C.RAY_ADDRESS).value == "127.0.0.1:6379"

    def test_rayservice_cluster_gets_sla_envs(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster, creator=C.KIND_RAYSERVICE)
        names = pod.spec.containers[0].env_names()
        assert C.RAY_TIMEOUT_MS_TASK_WAIT_FOR_DEATH_INFO in names
        assert C.RAY_GCS_SERVER_REQUEST_TIMEOUT_SECONDS in names

    def test_serve_label_for_rayservice_pods(self):
        cluster = simple_raycluster("demo")
        wpod = build_worker_pod(cluster, creator=C.KIND_RAYSERVICE)
        hpod = build_head_pod(cluster, creator=C.KIND_RAYSERVICE)
        assert wpod.metadata.labels[C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY] == "true"
        assert hpod.metadata.labels[C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY] == "false"


class TestTemplates:
    def test_labels(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        labels = pod.metadata.labels
        assert labels[C.RAY_CLUSTER_LABEL_KEY] == "demo"
        assert labels[C.RAY_NODE_TYPE_LABEL_KEY] == "worker"
        assert labels[C.RAY_NODE_GROUP_LABEL_KEY] == "default-group"
        assert labels[C.RAY_NODE_LABEL_KEY] == "yes"
        assert labels[C.KUBERNETES_CREATED_BY_LABEL_KEY] == C.COMPONENT_NAME

    def test_wait_gcs_ready_init_container(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        inits = pod.spec.init_containers
        assert inits and inits[0].name == "wait-gcs-ready"
        assert "ray health-check" in inits[0].args[0]
        assert inits[0].get_env(C.FQ_RAY_IP) is not None

    def test_shm_volume(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        vols = {v.name: v for v in pod.spec.volumes or []}
        assert C.SHARED_MEMORY_VOLUME_NAME in vols
        assert vols[C.SHARED_MEMORY_VOLUME_NAME].empty_dir["medium"] == "Memory"

    def test_autoscaler_sidecar_injected(self):
        cluster = simple_raycluster("demo", enableInTreeAutoscaling=True)
        pod = build_head_pod(cluster)
        names_ = [c.name for c in pod.spec.containers]
        assert "autoscaler" in names_
        # no-monitor param present in generated command
        assert "--no-monitor" in pod.spec.containers[0].args[0]

    def test_metrics_port_added(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        ports = {p.name: p.container_port for p in pod.spec.containers[0].ports or []}
        assert ports.get(C.METRICS_PORT_NAME) == C.DEFAULT_METRICS_PORT

    def test_gcs_ft_redis_env(self):
        cluster = simple_raycluster("demo", gcsFaultToleranceOptions={
            "backend": "redis", "redisAddress": "redis://r:6379"})
        hpod = build_head_pod(cluster)
        env = {e.name: e.value for e in hpod.spec.containers[0].env}
        assert env[C.RAY_REDIS_ADDRESS] == "redis://r:6379"
        assert C.RAY_EXTERNAL_STORAGE_NS in env
        wpod = build_worker_pod(cluster)
        wenv = {e.name: e.value for e in wpod.spec.containers[0].env}
        assert wenv[C.RAY_GCS_RPC_SERVER_RECONNECT_TIMEOUT_S] == "600"

    def test_tls_mounts(self):
        cluster = simple_raycluster("demo", tlsOptions={"enabled": True})
        pod = build_head_pod(cluster)
        env = {e.name: e.value for e in pod.spec.containers[0].env}
        assert env[C.RAY_USE_TLS] == "1"
        mounts = {m.name for m in pod.spec.containers[0].volume_mounts or []}
        assert C.RAY_TLS_VOLUME_NAME in mounts

    def test_probe_unified_http_for_new_ray(self):
        cluster = simple_raycluster("demo")
        cluster.spec.ray_version = "2.53.0"
        pod = build_worker_pod(cluster)
        probe = pod.spec.containers[0].readiness_probe
        assert probe.http_get is not None
        assert probe.http_get.port == C.DEFAULT_DASHBOARD_AGENT_LISTEN_PORT

    def test_probe_wget_for_old_ray(self):
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)  # 2.46.0
        probe = pod.spec.containers[0].readiness_probe
        assert probe.exec_ is not None and "wget" in probe.exec_.command[-1]


class TestServices:
    def test_head_service_ports_and_selector(self):
        cluster = simple_raycluster("demo")
        svc = servicelib.build_head_service(cluster)
        assert svc.metadata.name == "demo-head-svc"
        ports = {p.name: p.port for p in svc.spec.ports}
        assert ports == {"client": 10001, "dashboard": 8265, "gcs-server": 6379,
                         "metrics": 8080, "serve": 8000}
        assert svc.spec.selector[C.RAY_NODE_TYPE_LABEL_KEY] == "head"

    def test_head_service_custom_gcs_port(self):
        cluster = simple_raycluster("demo")
        cluster.spec.head_group_spec.ray_start_params["port"] = "6380"
        svc = servicelib.build_head_service(cluster)
        ports = {p.name: p.port for p in svc.spec.ports}
        assert ports["gcs-server"] == 6380

    def test_serve_service_selector_targets_healthy_proxies(self):
        cluster = simple_raycluster("demo")
        svc = servicelib.build_serve_service(cluster, cluster)
        assert svc.spec.selector[C.RAY_CLUSTER_SERVING_SERVICE_LABEL_KEY] == "true"
        assert svc.metadata.name == "demo-serve-svc"

    def test_headless_service(self):
        cluster = simple_raycluster("demo")
        svc = servicelib.build_headless_service(cluster)
        assert svc.spec.cluster_ip == "None"
        assert svc.spec.publish_not_ready_addresses is True
        assert svc.metadata.name == "demo-headless"


class TestEnvFlagToggles:
    def test_deterministic_head_pod_name(self, monkeypatch):
        monkeypatch.setenv(C.ENABLE_DETERMINISTIC_HEAD_POD_NAME, "true")
        cluster = simple_raycluster("demo")
        t = podlib.default_head_pod_template(cluster, cluster.spec.head_group_spec,
                                             "demo-head-", "6379")
        assert t.metadata.name == "demo-head-"
        assert t.metadata.generate_name is None

    def test_login_shell_flag(self, monkeypatch):
        monkeypatch.setenv(C.ENABLE_LOGIN_SHELL, "true")
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        assert pod.spec.containers[0].command == ["/bin/bash", "-lc", "--"]

    def test_probes_injection_disable(self, monkeypatch):
        monkeypatch.setenv(C.ENABLE_PROBES_INJECTION, "false")
        cluster = simple_raycluster("demo")
        pod = build_worker_pod(cluster)
        assert pod.spec.containers[0].readiness_probe is None
        assert pod.spec.containers[0].liveness_probe is None

    def test_init_container_injection_disable(self, monkeypatch):
        monkeypatch.setenv(C.ENABLE_INIT_CONTAINER_INJECTION, "false")
        cluster = simple_raycluster("demo")
        group = cluster.spec.worker_group_specs[0]
        t = podlib.default_worker_pod_template(
            cluster, group, "w-", "x.default.svc.cluster.local", "6379")
        assert not t.spec.init_containers
