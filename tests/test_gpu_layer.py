"""GPU layer tests. CPU-runnable parsing/decision tests plus @gpu tests that
exercise the real MI355X (rocm-smi + the gfx950 gpuhealth extension)."""
import time

import pytest

from kuberay_amd.gpu import autoscaler as asc
from kuberay_amd.gpu import rocm_smi, topology
from kuberay_amd.kube.client import InMemoryClient
from kuberay_amd.testing import simple_raycluster
from kuberay_amd.utils import constants as C

CANNED_SMI = {
    "card0": {"GPU use (%)": "85",
              "VRAM Total Memory (B)": "309237645312",
              "VRAM Total Used Memory (B)": "270000000000",
              "Temperature (Sensor junction) (C)": "78.0"},
    "card1": {"GPU use (%)": "5",
              "VRAM Total Memory (B)": "309237645312",
              "VRAM Total Used Memory (B)": "1000000"},
}

TOPO_8GPU = "=========== Link Type between two GPUs ===========\n" + \
    "       " + "   ".join(f"GPU{i}" for i in range(8)) + "\n" + \
    "\n".join(
        f"GPU{i}   " + "   ".join("0" if i == j else "XGMI" for j in range(8))
        for i in range(8))


class TestRocmSmiParse:
    def test_parse(self):
        stats = rocm_smi.parse_rocm_smi_json(CANNED_SMI)
        assert len(stats) == 2
        assert stats[0].utilization_pct == 85.0
        assert 0.85 < stats[0].vram_used_fraction < 0.9
        assert stats[0].temperature_c == 78.0
        assert stats[1].vram_used_fraction < 0.001

    def test_summary(self):
        stats = rocm_smi.parse_rocm_smi_json(CANNED_SMI)
        s = rocm_smi.node_gpu_summary(stats)
        assert s["gpu_count"] == 2
        assert s["avg_utilization_pct"] == 45.0
        assert s["max_vram_used_fraction"] > 0.85


class TestTopology:
    def test_full_island(self):
        t = topology.parse_showtopo_text(TOPO_8GPU)
        assert t.num_gpus == 8
        assert t.fully_connected()
        assert t.islands() == [set(range(8))]

    def test_split_islands(self):
        text = ("=========== Link Type between two GPUs ===========\n"
                "       GPU0   GPU1   GPU2   GPU3\n"
                "GPU0   0      XGMI   PCIE   PCIE\n"
                "GPU1   XGMI   0      PCIE   PCIE\n"
                "GPU2   PCIE   PCIE   0      XGMI\n"
                "GPU3   PCIE   PCIE   XGMI   0\n")
        t = topology.parse_showtopo_text(text)
        assert not t.fully_connected()
        assert sorted(map(sorted, t.islands())) == [[0, 1], [2, 3]]


class TestAutoscalerPolicy:
    def _setup(self, util, hbm, replicas=1):
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=replicas, gpus_per_worker=1)
        cluster.metadata.annotations = {asc.AMD_AUTOSCALER_ANNOTATION: "true"}
        client.create(cluster)
        clock = {"t": 0.0}
        telemetry = lambda: {"gpu_count": 8, "avg_utilization_pct": util,
                             "max_utilization_pct": util,
                             "avg_vram_used_fraction": hbm,
                             "max_vram_used_fraction": hbm}
        a = asc.MI355XAutoscaler(client, telemetry=telemetry,
                                 policy=asc.AutoscalerPolicy(
                                     up_stable_s=5, idle_timeout_s=10, cooldown_s=1),
                                 clock=lambda: clock["t"])
        return client, a, clock

    def _replicas(self, client):
        from kuberay_amd.models import RayCluster
        return client.get(RayCluster, "default", "demo") \
            .spec.worker_group_specs[0].replicas

    def test_scale_up_on_high_util_after_stability(self):
        client, a, clock = self._setup(util=95, hbm=0.2)
        assert a.step() == []           # starts the high window
        clock["t"] = 6.0
        decisions = a.step()
        assert decisions and "scale-up" in decisions[0]
        assert self._replicas(client) == 2

    def test_scale_up_on_hbm_pressure(self):
        client, a, clock = self._setup(util=10, hbm=0.95)
        a.step()
        clock["t"] = 6.0
        assert any("scale-up" in d for d in a.step())

    def test_no_scale_up_before_stability_window(self):
        client, a, clock = self._setup(util=95, hbm=0.2)
        a.step()
        clock["t"] = 2.0
        assert a.step() == []
        assert self._replicas(client) == 1

    def test_scale_down_names_victim(self):
        client, a, clock = self._setup(util=2, hbm=0.01, replicas=2)
        # create the worker pods the victim picker will inspect
        for i in range(2):
            client.server.create({
                "kind": "Pod",
                "metadata": {"name": f"demo-w{i}", "namespace": "default",
                             "labels": {C.RAY_CLUSTER_LABEL_KEY: "demo",
                                        C.RAY_NODE_TYPE_LABEL_KEY: "worker",
                                        C.RAY_NODE_GROUP_LABEL_KEY: "default-group"}}})
        a.step()
        clock["t"] = 11.0
        decisions = a.step()
        assert decisions and "scale-down" in decisions[0]
        from kuberay_amd.models import RayCluster
        g = client.get(RayCluster, "default", "demo").spec.worker_group_specs[0]
        assert g.replicas == 1
        assert g.scale_strategy.workers_to_delete  # victim named, not random

    def test_respects_max_replicas(self):
        client, a, clock = self._setup(util=95, hbm=0.9, replicas=8)
        a.step()
        clock["t"] = 100.0
        a.step()
        assert self._replicas(client) == 8

    def test_cooldown(self):
        client, a, clock = self._setup(util=95, hbm=0.2)
        a.step(); clock["t"] = 6.0; a.step()
        assert self._replicas(client) == 2
        clock["t"] = 6.4  # within cooldown
        a.step()
        assert self._replicas(client) == 2

    def test_opt_in_annotation_required(self):
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=1, gpus_per_worker=1)
        client.create(cluster)
        a = asc.MI355XAutoscaler(client, telemetry=lambda: {
            "avg_utilization_pct": 99, "max_vram_used_fraction": 0.99})
        assert a.step() == []

    def test_cpu_only_group_ignored(self):
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=1, gpus_per_worker=0)
        cluster.metadata.annotations = {asc.AMD_AUTOSCALER_ANNOTATION: "true"}
        client.create(cluster)
        a = asc.MI355XAutoscaler(client, telemetry=lambda: {
            "avg_utilization_pct": 99, "max_vram_used_fraction": 0.99},
            policy=asc.AutoscalerPolicy(up_stable_s=0))
        a.step(); a.step()
        from kuberay_amd.models import RayCluster
        assert client.get(RayCluster, "default", "demo") \
            .spec.worker_group_specs[0].replicas == 1


# ---------------------------------------------------------------------------
# real-GPU tests (gpurun)
# ---------------------------------------------------------------------------

@pytest.mark.gpu
class TestOnDevice:
    def test_rocm_smi_live(self):
        stats = rocm_smi.get_gpu_stats()
        assert len(stats) >= 1
        assert stats[0].vram_total_bytes > 100 * 1024**3  # 288 GB HBM3E

    def test_gpuhealth_extension_loads_and_identifies_gfx950(self):
        from kuberay_amd._native import gpuhealth
        assert gpuhealth.device_count() >= 1
        info = gpuhealth.device_info(0)
        assert "gfx950" in info["gcn_arch"]
        assert info["warp_size"] == 64

    def test_mfma_smoke(self):
        from kuberay_amd._native import gpuhealth
        assert gpuhealth.mfma_smoke(0, 2048)

    def test_hbm_bandwidth_sane(self):
        from kuberay_amd._native import gpuhealth
        bw = gpuhealth.hbm_bandwidth_gb_s(0, gib=2.0, iters=5)
        # healthy MI355X streams multiple TB/s; floor far below, cap far above
        assert 1000.0 < bw < 9000.0, bw

    def test_full_health_gate(self):
        from kuberay_amd.gpu.health import check_gpu_health
        report = check_gpu_health(quick=False)
        assert report.healthy, report
        assert report.mfma_ok
        assert report.hbm_gb_s > 1000

    def test_probe_cli(self):
        from kuberay_amd.gpu.probe import main
        assert main(["--quick", "--json"]) == 0

    def test_topology_live(self):
        t = topology.discover()
        assert t is not None
        assert t.num_gpus >= 1

    def test_gpu_gated_kubelet_marks_gpu_pod_ready(self):
        """End-to-end on device: a RayCluster with a GPU worker goes Ready
        only after the real MFMA/HBM health gate passes."""
        from kuberay_amd.gpu.health import sim_kubelet_gpu_gate
        from kuberay_amd.testing import ControlPlane
        cp = ControlPlane(kubelet_delay=0.01, gpu_gate=sim_kubelet_gpu_gate,
                          poll_seconds=0.05)
        cp.start()
        try:
            cp.client.create(simple_raycluster("gputest", workers=1,
                                               gpus_per_worker=1))
            assert cp.wait_cluster_state("default", "gputest", "ready",
                                         timeout=60)
        finally:
            cp.stop()


@pytest.mark.gpu
class TestAutoscalerOnDevice:
    def test_scale_down_from_real_idle_telemetry(self):
        """Real rocm-smi feed: an idle MI355X drives the scale-down path
        (BASELINE config #5's down direction, real telemetry)."""
        from kuberay_amd.gpu.rocm_smi import get_gpu_stats, node_gpu_summary
        from kuberay_amd.models import RayCluster
        from kuberay_amd.testing import ControlPlane, simple_raycluster

        stats = get_gpu_stats()
        summary = node_gpu_summary(stats)
        assert summary["gpu_count"] >= 1
        # the box should be idle during tests; if something else saturates
        # the GPU this test is not meaningful
        if summary["avg_utilization_pct"] > 50:
            pytest.skip("GPU busy; idle-telemetry scale-down not testable")

        cp = ControlPlane(kubelet_delay=0.01, poll_seconds=0.05)
        cp.start()
        try:
            cluster = simple_raycluster("realscale", workers=2, gpus_per_worker=1)
            cluster.metadata.annotations = {asc.AMD_AUTOSCALER_ANNOTATION: "true"}
            cp.client.create(cluster)
            assert cp.wait_for(
                lambda: cp.client.get(RayCluster, "default", "realscale")
                .status.available_worker_replicas == 2, timeout=30)
            autoscaler = asc.MI355XAutoscaler(
                cp.client, telemetry=node_gpu_summary,
                policy=asc.AutoscalerPolicy(idle_timeout_s=0.2, cooldown_s=0.1,
                                            down_util_pct=50,
                                            down_hbm_fraction=0.5))
            import time as _t
            deadline = _t.monotonic() + 30
            while _t.monotonic() < deadline:
                autoscaler.step()
                rc = cp.client.get(RayCluster, "default", "realscale")
                if rc.status.available_worker_replicas == 1:
                    break
                _t.sleep(0.1)
            rc = cp.client.get(RayCluster, "default", "realscale")
            assert rc.status.available_worker_replicas == 1
        finally:
            cp.stop()

    def test_metrics_observe_real_gpu_stats(self):
        from kuberay_amd.gpu.rocm_smi import get_gpu_stats
        from kuberay_amd.metrics import OperatorMetrics
        m = OperatorMetrics()
        m.observe_gpu_stats(get_gpu_stats())
        expo = m.exposition().decode()
        assert 'kuberay_mi355x_hbm_used_fraction{gpu="0"}' in expo


class TestNodeLabeller:
    def test_compute_labels_full_island(self):
        from kuberay_amd.gpu.labeller import compute_node_labels
        topo = topology.parse_showtopo_text(TOPO_8GPU)
        labels = compute_node_labels(topo, "mi355x-node-1")
        assert labels["amd.com/xgmi-island"] == "mi355x-node-1-island0"
        assert labels["amd.com/gpu.count"] == "8"
        assert labels["amd.com/xgmi-fully-connected"] == "true"

    def test_label_node_in_memory(self):
        from kuberay_amd.gpu.labeller import label_node
        client = InMemoryClient()
        client.server.create({"kind": "Node",
                              "metadata": {"name": "n1", "namespace": "default"}})
        topo = topology.parse_showtopo_text(TOPO_8GPU)
        labels = label_node(client, "n1", topo)
        assert labels
        node = client.server.get("Node", "default", "n1")
        assert node["metadata"]["labels"]["amd.com/xgmi-island"] == "n1-island0"

    def test_no_labels_without_gpus(self):
        from kuberay_amd.gpu.labeller import compute_node_labels
        assert compute_node_labels(None, "n1") == {}


class TestAutoscalerConflictTolerance:
    def test_conflict_during_update_skips_cluster(self):
        client = InMemoryClient()
        cluster = simple_raycluster("demo", workers=1, gpus_per_worker=1)
        cluster.metadata.annotations = {asc.AMD_AUTOSCALER_ANNOTATION: "true"}
        client.create(cluster)
        clock = {"t": 0.0}
        a = asc.MI355XAutoscaler(
            client,
            telemetry=lambda: {"avg_utilization_pct": 95,
                               "max_vram_used_fraction": 0.9, "gpu_count": 8},
            policy=asc.AutoscalerPolicy(up_stable_s=0, cooldown_s=0),
            clock=lambda: clock["t"])
        # sabotage: every update sees a concurrent writer bumping the rv
        original_update = client.update
        def racing_update(obj):
            from kuberay_amd.models import RayCluster as RC
            fresh = client.get(RC, "default", "demo")
            fresh.metadata.annotations["racer"] = str(clock["t"])
            original_update(fresh)
            return original_update(obj)  # now stale -> ConflictError
        client.update = racing_update
        clock["t"] = 1.0
        decisions = a.step()  # must not raise
        assert decisions == [] or decisions  # step survived
        client.update = original_update
        clock["t"] = 2.0
        assert any("scale-up" in d for d in a.step())


@pytest.mark.gpu
class TestBaselineConfigsOnDevice:
    """BASELINE.json configs #3 and #4 exercised against the real MI355X
    health gate: every GPU pod only turns Ready after the on-device
    MFMA/HBM probe passes."""

    def _plane(self):
        from kuberay_amd.gpu.health import sim_kubelet_gpu_gate
        from kuberay_amd.testing import ControlPlane
        return ControlPlane(kubelet_delay=0.01, gpu_gate=sim_kubelet_gpu_gate,
                            poll_seconds=0.05)

    def test_rayjob_ephemeral_8gpu_cluster_auto_deletes(self):
        """Config #3: RayJob submit -> ephemeral RayCluster with an
        amd.com/gpu=8 worker group, auto-delete on completion."""
        from kuberay_amd.models import RayJob
        from kuberay_amd.testing import simple_raycluster
        cp = self._plane()
        cp.start()
        try:
            spec = simple_raycluster("x", workers=1,
                                     gpus_per_worker=8).spec.to_dict()
            job = RayJob.from_dict({
                "apiVersion": "ray.io/v1", "kind": "RayJob",
                "metadata": {"name": "gpujob", "namespace": "default"},
                "spec": {"entrypoint": "python train.py",
                         "shutdownAfterJobFinishes": True,
                         "ttlSecondsAfterFinished": 0,
                         "rayClusterSpec": spec}})
            cp.client.create(job)
            assert cp.wait_for(
                lambda: cp.client.get(RayJob, "default", "gpujob")
                .status.job_deployment_status == "Complete", timeout=120)
            # ephemeral cluster auto-deleted after completion
            assert cp.wait_for(
                lambda: not cp.server.list("RayCluster", "default"),
                timeout=60)
        finally:
            cp.stop()

    def test_rayservice_zero_downtime_rollout_two_gpu_groups(self):
        """Config #4: RayService zero-downtime rollout across two MI355X
        worker groups (spec change -> pending cluster -> promote)."""
        from kuberay_amd.models import RayService
        from kuberay_amd.testing import simple_raycluster
        cp = self._plane()
        cp.start()
        try:
            cluster_spec = simple_raycluster(
                "x", workers=1, gpus_per_worker=1).spec.to_dict()
            cluster_spec["workerGroupSpecs"].append({
                **cluster_spec["workerGroupSpecs"][0],
                "groupName": "mi355x-b"})
            svc = RayService.from_dict({
                "apiVersion": "ray.io/v1", "kind": "RayService",
                "metadata": {"name": "gpusvc", "namespace": "default"},
                "spec": {"serveConfigV2":
                         "applications:\n- name: a\n  import_path: m.g\n",
                         "rayClusterConfig": cluster_spec}})
            cp.client.create(svc)

            def ready():
                s = cp.client.get(RayService, "default", "gpusvc")
                return s.status.service_status == "Running" and \
                    s.status.active_service_status.ray_cluster_name
            assert cp.wait_for(ready, timeout=120)
            first = cp.client.get(RayService, "default", "gpusvc") \
                .status.active_service_status.ray_cluster_name

            # zero-downtime upgrade: worker image change -> new cluster
            s = cp.client.get(RayService, "default", "gpusvc")
            s.spec.ray_cluster_spec.worker_group_specs[0].template.spec \
                .containers[0].image = "rocm/ray:2.47.0"
            cp.client.update(s)

            def promoted():
                cur = cp.client.get(RayService, "default", "gpusvc")
                active = cur.status.active_service_status.ray_cluster_name
                return active and active != first and \
                    cur.status.service_status == "Running"
            assert cp.wait_for(promoted, timeout=120)
        finally:
            cp.stop()


def test_probe_cli_fails_loudly_without_gpu():
    """On a non-GPU node the readiness probe must fail LOUDLY (exit != 0),
    never silently pass — a silent fallback would mark GPU pods Ready on
    nodes that can't run them."""
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, "-m", "kuberay_amd.gpu.probe", "--quick", "--json"],
        capture_output=True, text=True, timeout=60)
    if out.returncode == 0:
        import pytest
        pytest.skip("running on a GPU node")
    assert out.returncode != 0
    assert "not a GPU node" in out.stdout + out.stderr
