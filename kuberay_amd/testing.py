"""Test/bench harness: a full in-process control plane (the envtest analog).

Wires the in-memory API server + controller manager + simulated kubelet so
reconciler behavior can be exercised end-to-end without a real cluster —
the same seam the reference gets from controller-runtime envtest + the
FakeRayDashboardClient (ray-operator/controllers/ray/suite_test.go:57-120).
"""
from __future__ import annotations

import time
from typing import Callable, Dict, Optional

from .kube.client import InMemoryClient
from .kube.controller import Controller, Manager
from .kube.events import NullRecorder, StoreRecorder
from .kube.kubelet import SimKubelet
from .kube.store import InMemoryApiServer
from .models import RayCluster
from .utils import constants as C

OWNED_KINDS = ["Pod", "Service", "Secret", "PersistentVolumeClaim", "Job"]


class ControlPlane:
    def __init__(self, *, kubelet_delay: float = 0.0, job_runtime: float = 0.05,
                 workers: int = 4, record_events: bool = True,
                 gpu_gate: Optional[Callable[[dict], bool]] = None,
                 dashboard_client=None, enable_kubelet: bool = True,
                 requeue_seconds: Optional[int] = 300,
                 poll_seconds: Optional[float] = None,
                 server: Optional[InMemoryApiServer] = None,
                 kubelet_executors: int = 1):
        from .ops.raycluster import RayClusterReconciler, RayClusterReconcilerOptions
        from .ops.rayjob import RayJobReconciler
        from .ops.rayservice import RayServiceReconciler
        from .ops.raycronjob import RayCronJobReconciler
        from .utils.fake_dashboard import FakeRayDashboardClient

        # an existing server can be adopted (operator-restart tests resume
        # over surviving apiserver state)
        self.server = server or InMemoryApiServer()
        self.client = InMemoryClient(self.server)
        self.recorder = StoreRecorder(self.server) if record_events else NullRecorder()
        self.dashboard = dashboard_client or FakeRayDashboardClient()

        options = RayClusterReconcilerOptions()
        if requeue_seconds is not None:
            options.requeue_after_seconds = requeue_seconds
        self.raycluster_reconciler = RayClusterReconciler(
            self.client, recorder=self.recorder, options=options)
        self.rayjob_reconciler = RayJobReconciler(
            self.client, recorder=self.recorder, dashboard_factory=lambda url: self.dashboard)
        self.rayservice_reconciler = RayServiceReconciler(
            self.client, recorder=self.recorder, dashboard_factory=lambda url: self.dashboard)
        self.raycronjob_reconciler = RayCronJobReconciler(
            self.client, recorder=self.recorder)
        if poll_seconds is not None:
            self.rayjob_reconciler.requeue_seconds = poll_seconds
            self.rayservice_reconciler.requeue_seconds = poll_seconds

        self.manager = Manager(self.server)
        self.manager.add_controller(Controller(
            "raycluster", "RayCluster", self.raycluster_reconciler,
            owned_kinds=OWNED_KINDS, workers=workers))
        self.manager.add_controller(Controller(
            "rayjob", "RayJob", self.rayjob_reconciler,
            owned_kinds=["RayCluster", "Job"], workers=workers))
        self.manager.add_controller(Controller(
            "rayservice", "RayService", self.rayservice_reconciler,
            owned_kinds=["RayCluster", "Service"], workers=workers))
        self.manager.add_controller(Controller(
            "raycronjob", "RayCronJob", self.raycronjob_reconciler,
            owned_kinds=["RayJob"], workers=1))

        self.kubelet = (SimKubelet(self.server, startup_delay=kubelet_delay,
                                   job_runtime=job_runtime, gpu_gate=gpu_gate,
                                   executors=kubelet_executors)
                        if enable_kubelet else None)

    # -- lifecycle -----------------------------------------------------
    def start(self) -> "ControlPlane":
        self.manager.start()
        if self.kubelet:
            self.kubelet.start()
        return self

    def stop(self) -> None:
        if self.kubelet:
            self.kubelet.stop()
        self.manager.stop()

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    # -- helpers -------------------------------------------------------
    def wait_for(self, predicate: Callable[[], bool], timeout: float = 10.0,
                 interval: float = 0.02) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if predicate():
                return True
            time.sleep(interval)
        return predicate()

    def wait_cluster_state(self, namespace: str, name: str, state: str,
                           timeout: float = 10.0) -> bool:
        def check():
            rc = self.client.try_get(RayCluster, namespace, name)
            return rc is not None and rc.status.state == state
        return self.wait_for(check, timeout)

    def wait_cluster_condition(self, namespace: str, name: str, cond: str,
                               status: str = "True", timeout: float = 10.0) -> bool:
        def check():
            rc = self.client.try_get(RayCluster, namespace, name)
            if rc is None:
                return False
            for c in rc.status.conditions or []:
                if c.type == cond and c.status == status:
                    return True
            return False
        return self.wait_for(check, timeout)


def simple_raycluster(name: str = "test-cluster", namespace: str = "default",
                      workers: int = 1, gpus_per_worker: int = 0,
                      num_of_hosts: int = 1, **spec_overrides) -> RayCluster:
    """A minimal valid RayCluster matching the reference sample YAMLs."""
    worker_limits: Dict[str, str] = {"cpu": "1", "memory": "1Gi"}
    if gpus_per_worker:
        worker_limits[C.AMD_GPU_RESOURCE_NAME] = str(gpus_per_worker)
    spec = {
        "rayVersion": "2.46.0",
        "headGroupSpec": {
            "rayStartParams": {},
            "template": {"spec": {"containers": [{
                "name": "ray-head",
                "image": C.DEFAULT_RAY_ROCM_IMAGE,
                "resources": {"limits": {"cpu": "1", "memory": "2Gi"}},
            }]}},
        },
        "workerGroupSpecs": [{
            "groupName": "default-group",
            "replicas": workers,
            "minReplicas": 0,
            "maxReplicas": max(workers, 8),
            "numOfHosts": num_of_hosts,
            "rayStartParams": {},
            "template": {"spec": {"containers": [{
                "name": "ray-worker",
                "image": C.DEFAULT_RAY_ROCM_IMAGE,
                "resources": {"limits": worker_limits},
            }]}},
        }] if workers is not None else [],
    }
    spec.update(spec_overrides)
    return RayCluster.from_dict({
        "apiVersion": C.API_VERSION,
        "kind": C.KIND_RAYCLUSTER,
        "metadata": {"name": name, "namespace": namespace},
        "spec": spec,
    })
