"""Operator binary (reference: ray-operator/main.go:59-433).

``python -m kuberay_amd.operator [flags]`` builds the controller manager
(all four CR reconcilers + networkpolicy when gated), wires metrics,
health endpoints, the batch scheduler and the MI355X autoscaler, and runs
until SIGTERM.

Backends:
  * ``memory``      — self-contained control plane (in-memory apiserver +
                      simulated kubelet); used for local runs and soak tests,
  * ``kubernetes``  — a real cluster through kuberay_amd.kube.rest.
"""
from __future__ import annotations

import logging
import os
import signal
import sys
import threading

from . import features
from .config import Configuration, load_config
from .kube.client import InMemoryClient
from .kube.controller import Controller, Manager
from .kube.events import StoreRecorder
from .kube.store import InMemoryApiServer
from .metrics import OperatorMetrics

logger = logging.getLogger("kuberay.operator")


def build_manager(cfg: Configuration, server=None, client=None):
    from .ops.raycluster import RayClusterReconciler, RayClusterReconcilerOptions
    from .ops.rayjob import RayJobReconciler
    from .ops.rayservice import RayServiceReconciler
    from .ops.raycronjob import RayCronJobReconciler
    from .ops.networkpolicy import NetworkPolicyReconciler
    from .parallel import scheduler_for

    features.parse_feature_gates(cfg.feature_gates)

    metrics = OperatorMetrics() if cfg.enable_metrics else None
    if server is None:
        if cfg.backend == "memory":
            server = InMemoryApiServer()
        else:
            from .kube.rest import RestApiServerAdapter
            server = RestApiServerAdapter(kubeconfig=cfg.kubeconfig,
                                          metrics=metrics)
    if client is None:
        client = InMemoryClient(server) if isinstance(server, InMemoryApiServer) else server.client()

    recorder = StoreRecorder(server) if isinstance(server, InMemoryApiServer) else None

    scheduler = None
    if features.enabled("KubernetesWAS"):
        # the KubernetesWAS gate SELECTS the WAS scheduler and must not be
        # combined with --batch-scheduler/--enable-batch-scheduler
        # (reference: schedulermanager.go:56-60 + ValidateBatchSchedulerConfig)
        if cfg.enable_batch_scheduler or cfg.batch_scheduler:
            raise ValueError(
                "feature gate KubernetesWAS cannot be combined with "
                "--enable-batch-scheduler/--batch-scheduler")
        scheduler = scheduler_for("kubernetes-was-v1alpha2")
    elif cfg.enable_batch_scheduler:
        name = cfg.batch_scheduler or (
            "xgmi-gang" if features.enabled("XgmiGangScheduling") else None)
        scheduler = scheduler_for(name)

    options = RayClusterReconcilerOptions()
    from .kube.objects import Container
    for c in cfg.head_sidecar_containers or []:
        options.head_sidecar_containers.append(Container.from_dict(c))
    for c in cfg.worker_sidecar_containers or []:
        options.worker_sidecar_containers.append(Container.from_dict(c))

    manager = Manager(server)
    workers = cfg.reconcile_concurrency
    ns_scope = cfg.watch_namespaces or None
    shard = (cfg.shard_index, cfg.shards) if cfg.shards > 1 else None
    manager.add_controller(Controller(
        "raycluster", "RayCluster",
        RayClusterReconciler(client, recorder=recorder, batch_scheduler=scheduler,
                             options=options, metrics=metrics),
        owned_kinds=["Pod", "Service", "Secret", "PersistentVolumeClaim", "Job"],
        workers=workers, watch_namespaces=ns_scope, metrics=metrics,
        shard=shard))
    manager.add_controller(Controller(
        "rayjob", "RayJob",
        RayJobReconciler(client, recorder=recorder, metrics=metrics),
        owned_kinds=["RayCluster", "Job"], workers=workers,
        watch_namespaces=ns_scope, metrics=metrics, shard=shard))
    # real clusters: serve-proxy healthz drives the head pod's serve label
    http_proxy = None
    if cfg.backend == "kubernetes":
        from .utils.dashboard_client import RayHttpProxyClient
        http_proxy = RayHttpProxyClient()
    manager.add_controller(Controller(
        "rayservice", "RayService",
        RayServiceReconciler(client, recorder=recorder, metrics=metrics,
                             http_proxy_client=http_proxy),
        owned_kinds=["RayCluster", "Service"], workers=workers,
        watch_namespaces=ns_scope, metrics=metrics, shard=shard))
    if features.enabled("RayCronJob"):
        manager.add_controller(Controller(
            "raycronjob", "RayCronJob",
            RayCronJobReconciler(client, recorder=recorder),
            owned_kinds=["RayJob"], workers=1, shard=shard))
    if features.enabled("RayClusterNetworkPolicy"):
        manager.add_controller(Controller(
            "networkpolicy", "RayCluster",
            NetworkPolicyReconciler(client, recorder=recorder),
            owned_kinds=["NetworkPolicy"], workers=1, shard=shard))
    if features.enabled("RayClusterMTLS"):
        from .ops.mtls import MTLSReconciler
        manager.add_controller(Controller(
            "mtls", "RayCluster", MTLSReconciler(client, recorder=recorder),
            owned_kinds=["Secret", "Pod"], workers=1, shard=shard))

    autoscaler = None
    if cfg.enable_mi355x_autoscaler and features.enabled("MI355XAutoscaler"):
        from .gpu.autoscaler import MI355XAutoscaler
        autoscaler = MI355XAutoscaler(client, recorder=recorder,
                                      metrics=metrics)

    return manager, client, metrics, autoscaler


class HealthServer(threading.Thread):
    """/healthz /readyz /metrics on probe/metrics addrs (single port here)."""

    def __init__(self, manager: Manager, metrics, port: int):
        super().__init__(daemon=True, name="health-server")
        self.manager = manager
        self.metrics = metrics
        self.port = port
        self._httpd = None

    def run(self) -> None:
        import http.server

        manager, metrics = self.manager, self.metrics

        class Handler(http.server.BaseHTTPRequestHandler):
            def do_GET(self):  # noqa: N802
                if self.path in ("/healthz", "/readyz"):
                    self.send_response(200)
                    self.end_headers()
                    self.wfile.write(b"ok")
                elif self.path == "/metrics" and metrics is not None:
                    body = metrics.exposition()
                    self.send_response(200)
                    self.send_header("Content-Type", "text/plain; version=0.0.4")
                    self.end_headers()
                    self.wfile.write(body)
                else:
                    self.send_response(404)
                    self.end_headers()

            def log_message(self, *a):
                pass

        try:
            self._httpd = http.server.ThreadingHTTPServer(("0.0.0.0", self.port), Handler)
        except OSError as e:
            logger.warning("health server bind failed: %s", e)
            return
        self._httpd.serve_forever()

    def stop(self) -> None:
        if self._httpd:
            self._httpd.shutdown()


def main(argv=None) -> int:
    logging.basicConfig(
        level=logging.INFO,
        format='{"ts":"%(asctime)s","level":"%(levelname)s",'
               '"logger":"%(name)s","msg":"%(message)s"}')
    cfg = load_config(argv)
    manager, client, metrics, autoscaler = build_manager(cfg)

    port = int(cfg.metrics_addr.rsplit(":", 1)[-1] or 8080)
    health = HealthServer(manager, metrics, port)
    health.start()

    kubelet = None
    snapshotter = None
    facade = None
    if cfg.backend == "memory":
        if cfg.api_port:
            from .kube.httpserver import KubeApiFacade
            facade = KubeApiFacade(manager.server, port=cfg.api_port).start()
            logger.info("kube-API facade serving at %s", facade.url)
        if cfg.state_file:
            from .kube.snapshot import SnapshotLoop, load_snapshot
            restored = load_snapshot(manager.server, cfg.state_file)
            if restored:
                logger.info("restored %d objects from %s", restored,
                            cfg.state_file)
            snapshotter = SnapshotLoop(manager.server, cfg.state_file,
                                       cfg.state_snapshot_interval_s)
            snapshotter.start()
        from .kube.kubelet import SimKubelet
        gate = None
        if features.enabled("MI355XGpuHealthProbes"):
            from .gpu.health import gpu_node, sim_kubelet_gpu_gate
            if gpu_node():
                gate = sim_kubelet_gpu_gate
        kubelet = SimKubelet(manager.server, gpu_gate=gate)
        kubelet.start()

    elector = None
    if cfg.enable_leader_election and cfg.backend == "kubernetes":
        from .kube.leaderelection import LeaderElector
        import socket
        started = threading.Event()
        lease = "kuberay-amd-operator" if cfg.shards <= 1 else \
            f"kuberay-amd-operator-shard-{cfg.shard_index}"
        elector = LeaderElector(
            client,
            lease_name=lease,
            namespace=cfg.leader_election_namespace or "ray-system",
            identity=f"{socket.gethostname()}-{os.getpid()}",
            lease_duration_s=cfg.leader_lease_seconds,
            on_started_leading=started.set)
        elector.start()
        logger.info("waiting for leader election (identity=%s)",
                    elector.identity)
        started.wait()
    manager.start()
    if autoscaler is not None:
        autoscaler.start(cfg.mi355x_autoscaler_interval_s)
    logger.info("kuberay-amd operator started (backend=%s, gates=%s)",
                cfg.backend, features.all_gates())

    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *a: stop.set())
    signal.signal(signal.SIGINT, lambda *a: stop.set())
    try:
        while not stop.is_set():
            stop.wait(1.0)
    finally:
        if autoscaler is not None:
            autoscaler.stop()
        if kubelet is not None:
            kubelet.stop()
        if snapshotter is not None:
            snapshotter.stop()
        if facade is not None:
            facade.stop()
        if elector is not None:
            elector.stop()
        manager.stop()
        health.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
