"""Simulated kubelet for tests and the scale bench.

The reference's envtest suites run reconcilers against an apiserver with NO
kubelet and hand-set pod statuses (suite_test.go; note at
raycluster_controller.go:1575-1578). This module automates that hand-setting
with realistic lifecycle semantics so the 500-cluster soak measures the
operator, not test plumbing:

* ADDED pods transition Pending -> Running(+Ready) after ``startup_delay``,
* each pod gets a synthetic IP,
* batch Jobs complete after ``job_runtime``,
* optional ``gpu_gate`` hook: called before a GPU pod turns Ready — on a
  real MI355X box this is wired to the rocm-smi / on-device MFMA health
  probe (kuberay_amd.gpu.health), making readiness genuinely GPU-backed.
"""
from __future__ import annotations

import heapq
import itertools
import threading
import time
from typing import Callable, List, Optional, Tuple

from ..utils import constants as C
from .store import InMemoryApiServer, NotFoundError, now_iso


class SimKubelet:
    def __init__(
        self,
        server: InMemoryApiServer,
        startup_delay: float = 0.0,
        job_runtime: float = 0.05,
        gpu_gate: Optional[Callable[[dict], bool]] = None,
        executors: int = 1,
    ):
        self.server = server
        self.startup_delay = startup_delay
        self.job_runtime = job_runtime
        self.gpu_gate = gpu_gate
        # a real cluster has one kubelet PER NODE acting in parallel, so
        # due timers can drain onto an executor pool; measured on the
        # 256-core MI355X box extra timer threads CONTEND on the GIL with
        # the controller workers (258 -> 182 clusters/s at executors=4),
        # so the default stays 1 there while smaller boxes benefit
        self.executors = max(1, executors)
        self._ip_counter = itertools.count(1)
        self._timer_lock = threading.Condition()
        self._timers: List[Tuple[float, int, Callable[[], None]]] = []
        self._seq = 0
        self._stopped = threading.Event()
        self._threads: List[threading.Thread] = []
        self._watcher = None

    # -- lifecycle -----------------------------------------------------
    WATCHED_KINDS = ("Pod", "Job", "Gateway", "HTTPRoute")

    def start(self) -> None:
        self._watcher = self.server.watch(set(self.WATCHED_KINDS))
        for kind in self.WATCHED_KINDS:
            for obj in self.server.list(kind):
                self._on_event("ADDED", obj)
        t1 = threading.Thread(target=self._watch_loop, name="sim-kubelet-watch", daemon=True)
        self._threads = [t1]
        for i in range(self.executors):
            self._threads.append(threading.Thread(
                target=self._timer_loop, name=f"sim-kubelet-timer-{i}",
                daemon=True))
        for t in self._threads:
            t.start()

    def stop(self) -> None:
        self._stopped.set()
        if self._watcher:
            self._watcher.stop()
        with self._timer_lock:
            self._timer_lock.notify_all()
        for t in self._threads:
            t.join(timeout=2)

    # -- internals -----------------------------------------------------
    def _schedule(self, delay: float, fn: Callable[[], None]) -> None:
        with self._timer_lock:
            self._seq += 1
            heapq.heappush(self._timers, (time.monotonic() + delay, self._seq, fn))
            self._timer_lock.notify()

    def _timer_loop(self) -> None:
        while not self._stopped.is_set():
            with self._timer_lock:
                if not self._timers:
                    self._timer_lock.wait(0.2)
                    continue
                when, _, fn = self._timers[0]
                delta = when - time.monotonic()
                if delta > 0:
                    self._timer_lock.wait(min(delta, 0.2))
                    continue
                heapq.heappop(self._timers)
            try:
                fn()
            except NotFoundError:
                pass
            except Exception:
                pass

    def _watch_loop(self) -> None:
        while not self._stopped.is_set():
            ev = self._watcher.next(timeout=0.5)
            if ev is None:
                continue
            self._on_event(*ev)

    def _on_event(self, event_type: str, obj: dict) -> None:
        kind = obj.get("kind")
        if event_type != "ADDED":
            return
        meta = obj["metadata"]
        ns, name = meta.get("namespace", "default"), meta["name"]
        if kind == "Pod":
            self._schedule(self.startup_delay, lambda: self._start_pod(ns, name))
        elif kind == "Job":
            self._schedule(self.job_runtime, lambda: self._complete_job(ns, name))
        elif kind == "Gateway":
            self._schedule(self.startup_delay,
                           lambda: self._program_gateway(ns, name))
        elif kind == "HTTPRoute":
            self._schedule(self.startup_delay,
                           lambda: self._accept_route(ns, name))

    def _running_status(self, pod: dict) -> dict:
        ip = f"10.244.{next(self._ip_counter) % 255}.{next(self._ip_counter) % 255}"
        containers = pod.get("spec", {}).get("containers", [])
        return {
            "phase": "Running",
            "podIP": ip,
            "startTime": now_iso(),
            "conditions": [
                {"type": "PodScheduled", "status": "True"},
                {"type": "Initialized", "status": "True"},
                {"type": "ContainersReady", "status": "True"},
                {"type": "Ready", "status": "True",
                 "lastTransitionTime": now_iso()},
            ],
            "containerStatuses": [
                {"name": c.get("name", f"c{i}"), "ready": True,
                 "restartCount": 0,
                 "state": {"running": {"startedAt": now_iso()}}}
                for i, c in enumerate(containers)
            ],
        }

    def _start_pod(self, namespace: str, name: str) -> None:
        if self.gpu_gate is None:
            # hot path: one fetch + one status write under the store's lock
            def to_running(pod):
                if pod["metadata"].get("deletionTimestamp"):
                    return False
                pod["status"] = {**(pod.get("status") or {}),
                                 **self._running_status(pod)}
                return None
            self.server.mutate_status("Pod", namespace, name, to_running)
            return
        # gated path: the GPU health probe can be slow (rocm-smi / on-device
        # MFMA), so it must run OUTSIDE the store lock
        pod = self.server.try_get("Pod", namespace, name)
        if pod is None or pod["metadata"].get("deletionTimestamp"):
            return
        if self._requests_gpu(pod) and not self.gpu_gate(pod):
            # GPU unhealthy: pod stays Pending (readiness probe failing)
            self.server.patch_merge("Pod", namespace, name, {
                "status": {"phase": "Pending",
                           "reason": "GPUHealthCheckFailed"}},
                subresource="status")
            return
        self.server.patch_merge("Pod", namespace, name,
                                {"status": self._running_status(pod)},
                                subresource="status")

    @staticmethod
    def _requests_gpu(pod: dict) -> bool:
        for c in pod.get("spec", {}).get("containers", []):
            res = c.get("resources") or {}
            for block in (res.get("limits") or {}, res.get("requests") or {}):
                if C.AMD_GPU_RESOURCE_NAME in block:
                    return True
        return False

    def _program_gateway(self, namespace: str, name: str) -> None:
        """Stand-in gateway controller: accept + program every Gateway, so
        the incremental-upgrade readiness gate (Gateway API conditions) is
        genuinely exercised in envtest-style runs."""
        gw = self.server.try_get("Gateway", namespace, name)
        if gw is None or gw["metadata"].get("deletionTimestamp"):
            return
        self.server.patch_merge("Gateway", namespace, name, {
            "status": {"conditions": [
                {"type": "Accepted", "status": "True",
                 "lastTransitionTime": now_iso()},
                {"type": "Programmed", "status": "True",
                 "lastTransitionTime": now_iso()},
            ]}}, subresource="status")

    def _accept_route(self, namespace: str, name: str) -> None:
        route = self.server.try_get("HTTPRoute", namespace, name)
        if route is None or route["metadata"].get("deletionTimestamp"):
            return
        parents = []
        for ref in (route.get("spec") or {}).get("parentRefs") or []:
            parents.append({
                "parentRef": dict(ref),
                "controllerName": "kuberay-amd.sim/gateway-controller",
                "conditions": [
                    {"type": "Accepted", "status": "True",
                     "lastTransitionTime": now_iso()},
                    {"type": "ResolvedRefs", "status": "True",
                     "lastTransitionTime": now_iso()},
                ]})
        self.server.patch_merge("HTTPRoute", namespace, name,
                                {"status": {"parents": parents}},
                                subresource="status")

    def _complete_job(self, namespace: str, name: str) -> None:
        job = self.server.try_get("Job", namespace, name)
        if job is None or job["metadata"].get("deletionTimestamp"):
            return
        self.server.patch_merge("Job", namespace, name, {
            "status": {
                "succeeded": 1,
                "completionTime": now_iso(),
                "conditions": [{"type": "Complete", "status": "True"}],
            }
        }, subresource="status")
