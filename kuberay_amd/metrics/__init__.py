"""Prometheus metrics (reference: ray-operator/controllers/ray/metrics/).

Same exposition names as the reference so existing Grafana dashboards and
PodMonitor configs work unchanged:
  kuberay_cluster_provisioned_duration_seconds (ray_cluster_metrics.go:35-47)
  kuberay_cluster_info (:48-53)
  kuberay_cluster_condition_provisioned (:54-56)
  kuberay_job_execution_duration_seconds / kuberay_job_info /
  kuberay_job_deployment_status (ray_job_metrics.go:33-50)
  kuberay_service_info / kuberay_service_condition_* (ray_service_metrics.go)
plus control-plane internals (reconcile latency/count — the client-go
histogram analog, client_go_metrics.go:27-76).
"""
from __future__ import annotations

import time
from typing import Dict, Optional

from prometheus_client import (
    CollectorRegistry,
    Counter,
    Gauge,
    Histogram,
    REGISTRY,
    generate_latest,
)


class OperatorMetrics:
    def __init__(self, registry: Optional[CollectorRegistry] = None):
        self.registry = registry or CollectorRegistry()
        r = self.registry

        # -- RayCluster -------------------------------------------------
        self.cluster_provisioned_duration = Histogram(
            "kuberay_cluster_provisioned_duration_seconds",
            "Time from RayCluster creation until RayClusterProvisioned",
            ["namespace"], registry=r,
            buckets=(0.5, 1, 2, 5, 10, 30, 60, 120, 300, 600, 1800))
        self.cluster_info = Gauge(
            "kuberay_cluster_info", "RayCluster metadata",
            ["namespace", "name", "owner_kind"], registry=r)
        self.cluster_condition_provisioned = Gauge(
            "kuberay_cluster_condition_provisioned",
            "RayClusterProvisioned condition (1=true)",
            ["namespace", "name", "condition"], registry=r)

        # -- RayJob -----------------------------------------------------
        self.job_execution_duration = Histogram(
            "kuberay_job_execution_duration_seconds",
            "RayJob execution duration (start to terminal)",
            ["namespace", "job_deployment_status", "retry_count"], registry=r,
            buckets=(1, 5, 30, 60, 300, 900, 1800, 3600, 7200, 21600))
        self.job_info = Gauge(
            "kuberay_job_info", "RayJob metadata",
            ["namespace", "name"], registry=r)
        self.job_deployment_status = Gauge(
            "kuberay_job_deployment_status", "RayJob deployment status",
            ["namespace", "name", "deployment_status"], registry=r)

        # -- RayService -------------------------------------------------
        self.service_info = Gauge(
            "kuberay_service_info", "RayService metadata",
            ["namespace", "name"], registry=r)
        self.service_condition_ready = Gauge(
            "kuberay_service_condition_ready",
            "RayService Ready condition (1=true)",
            ["namespace", "name"], registry=r)
        self.service_condition_upgrade_in_progress = Gauge(
            "kuberay_service_condition_upgrade_in_progress",
            "RayService UpgradeInProgress condition (1=true)",
            ["namespace", "name"], registry=r)

        # -- control-plane internals (client-go histogram analog) -------
        self.reconcile_duration = Histogram(
            "kuberay_reconcile_duration_seconds",
            "Reconcile latency per controller", ["controller"], registry=r,
            buckets=(0.0005, 0.001, 0.005, 0.01, 0.05, 0.1, 0.5, 1, 5))
        self.reconcile_total = Counter(
            "kuberay_reconcile_total", "Reconciles per controller and outcome",
            ["controller", "outcome"], registry=r)
        self.workqueue_depth = Gauge(
            "kuberay_workqueue_depth", "Work queue depth", ["controller"],
            registry=r)

        # client request latency (reference: metrics/client_go_metrics.go
        # rest-client duration histograms)
        self.api_request_duration = Histogram(
            "kuberay_api_request_duration_seconds",
            "Kube API request latency by HTTP verb and status code",
            ["verb", "code"], registry=r)

        # -- MI355X-native ----------------------------------------------
        self.gpu_utilization = Gauge(
            "kuberay_mi355x_gpu_utilization_pct",
            "rocm-smi GPU utilization per device", ["gpu"], registry=r)
        self.gpu_hbm_used_fraction = Gauge(
            "kuberay_mi355x_hbm_used_fraction",
            "rocm-smi HBM (288GB) occupancy fraction per device", ["gpu"],
            registry=r)
        self.autoscaler_decisions = Counter(
            "kuberay_mi355x_autoscaler_decisions_total",
            "rocm-smi-driven autoscaler decisions", ["direction"], registry=r)
        self.gpu_health = Gauge(
            "kuberay_mi355x_gpu_health",
            "1 when the node's GPU telemetry/health probe succeeds, 0 when "
            "it fails (feeds the MI355XGpuUnhealthy alert rule)", ["node"],
            registry=r)

    # -- hooks used by the reconcilers ----------------------------------
    def observe_cluster_ready(self, cluster) -> None:
        import calendar
        created = cluster.metadata.creation_timestamp
        if created:
            try:
                t0 = calendar.timegm(time.strptime(created, "%Y-%m-%dT%H:%M:%SZ"))
                self.cluster_provisioned_duration.labels(
                    cluster.metadata.namespace or "default").observe(
                        max(0.0, time.time() - t0))
            except ValueError:
                pass
        owner = (cluster.metadata.labels or {}).get(
            "ray.io/originated-from-crd", "RayCluster")
        self.cluster_info.labels(cluster.metadata.namespace or "default",
                                 cluster.metadata.name, owner).set(1)
        self.cluster_condition_provisioned.labels(
            cluster.metadata.namespace or "default", cluster.metadata.name,
            "true").set(1)

    def observe_job_finished(self, rayjob, succeeded: bool) -> None:
        import calendar
        status = "Complete" if succeeded else "Failed"
        start = rayjob.status.start_time
        dur = 0.0
        if start:
            try:
                t0 = calendar.timegm(time.strptime(start, "%Y-%m-%dT%H:%M:%SZ"))
                dur = max(0.0, time.time() - t0)
            except ValueError:
                pass
        retries = (rayjob.status.failed or 0) + (rayjob.status.succeeded or 0) - 1
        self.job_execution_duration.labels(
            rayjob.metadata.namespace or "default", status,
            str(max(retries, 0))).observe(dur)
        self.job_deployment_status.labels(
            rayjob.metadata.namespace or "default", rayjob.metadata.name,
            status).set(1)

    def observe_gpu_stats(self, stats) -> None:
        for s in stats:
            self.gpu_utilization.labels(str(s.index)).set(s.utilization_pct)
            self.gpu_hbm_used_fraction.labels(str(s.index)).set(s.vram_used_fraction)

    def observe_gpu_health(self, node: str, healthy: bool) -> None:
        self.gpu_health.labels(node).set(1 if healthy else 0)

    def exposition(self) -> bytes:
        return generate_latest(self.registry)
