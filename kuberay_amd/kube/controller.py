"""Controller runtime: manager + controllers + reconcile loop.

The controller-runtime analog (reference wiring:
ray-operator/controllers/ray/raycluster_controller.go:1941-1968
SetupWithManager and ray-operator/main.go:308-369):

* one watch stream over the API server feeds all controllers,
* a controller enqueues its primary kind's keys, and maps owned kinds
  (Pod/Service/Secret/PVC/Job) to the owning CR via ownerReferences,
* MODIFIED events on the primary kind pass a generation/labels/annotations/
  finalizers predicate so status-only writes don't self-trigger
  (controller-runtime's GenerationChangedPredicate analog),
* per-key serialization + coalescing via the rate-limited workqueue,
* N worker threads per controller (reference default concurrency 1;
  apis/config/v1alpha1/defaults.go:8-13 — ours defaults higher since the
  Python reconcilers are pure functions over the in-memory cache).
"""
from __future__ import annotations

import dataclasses
import logging
import threading
import traceback
from typing import Any, Dict, List, Optional, Tuple

from .store import ConflictError, InMemoryApiServer
from .workqueue import RateLimitingQueue

logger = logging.getLogger("kuberay.controller")

Request = Tuple[str, str]  # (namespace, name)


@dataclasses.dataclass
class Result:
    requeue: bool = False
    requeue_after: Optional[float] = None  # seconds


class Reconciler:
    """Interface: reconcile((namespace, name)) -> Result."""

    def reconcile(self, request: Request) -> Result:  # pragma: no cover
        raise NotImplementedError


def _meta_fingerprint(obj: Dict[str, Any]) -> Tuple:
    meta = obj.get("metadata", {})
    return (
        meta.get("generation"),
        tuple(sorted((meta.get("labels") or {}).items())),
        tuple(sorted((meta.get("annotations") or {}).items())),
        meta.get("deletionTimestamp"),
        tuple(meta.get("finalizers") or []),
    )


class Controller:
    def __init__(
        self,
        name: str,
        primary_kind: str,
        reconciler: Reconciler,
        owned_kinds: Optional[List[str]] = None,
        workers: int = 4,
        use_predicates: bool = True,
        watch_namespaces: Optional[List[str]] = None,
        metrics=None,
        owned_event_coalesce_s: float = 0.0,
        shard: Optional[Tuple[int, int]] = None,
    ):
        self.name = name
        self.primary_kind = primary_kind
        self.reconciler = reconciler
        self.owned_kinds = set(owned_kinds or [])
        self.workers = workers
        self.use_predicates = use_predicates
        # informer scoping (reference: internal/managercache/cache.go:19-39)
        self.watch_namespaces = set(watch_namespaces) if watch_namespaces else None
        self.queue = RateLimitingQueue()
        self._fingerprints: Dict[Request, Tuple] = {}
        self._threads: List[threading.Thread] = []
        self._stopped = threading.Event()
        self.reconcile_count = 0
        self.error_count = 0
        self.metrics = metrics
        # optional delay on owned-object events to fold bursts into one
        # reconcile; 0 (default) measured fastest — the dedup queue already
        # coalesces while workers are busy
        self.owned_event_coalesce_s = owned_event_coalesce_s
        # horizontal sharding: (index, total) — this controller only handles
        # CRs whose stable name-hash lands on its shard, so N operator
        # processes can split one cluster's load (each shard elects its own
        # leader Lease). None = unsharded.
        self.shard = shard
        if shard is not None:
            index, total = shard
            if not (total >= 1 and 0 <= index < total):
                raise ValueError(f"invalid shard {shard}")

    def _owns(self, namespace: str, name: str) -> bool:
        if self.shard is None:
            return True
        index, total = self.shard
        # crc32: stable across processes (python str hash is randomized)
        import zlib
        return zlib.crc32(f"{namespace}/{name}".encode()) % total == index

    # -- event routing -------------------------------------------------
    def observe(self, event_type: str, obj: Dict[str, Any]) -> None:
        kind = obj.get("kind")
        meta = obj.get("metadata", {})
        if self.watch_namespaces is not None and \
                meta.get("namespace", "default") not in self.watch_namespaces:
            return
        if kind == self.primary_kind:
            key: Request = (meta.get("namespace", "default"), meta.get("name", ""))
            if not self._owns(*key):
                return
            if event_type == "MODIFIED" and self.use_predicates:
                fp = _meta_fingerprint(obj)
                if self._fingerprints.get(key) == fp:
                    return  # status-only update; don't self-trigger
                self._fingerprints[key] = fp
            elif event_type == "ADDED":
                self._fingerprints[key] = _meta_fingerprint(obj)
            elif event_type == "DELETED":
                self._fingerprints.pop(key, None)
            self.queue.add(key)
        elif kind in self.owned_kinds:
            for ref in meta.get("ownerReferences") or []:
                if ref.get("kind") == self.primary_kind:
                    key = (meta.get("namespace", "default"), ref.get("name", ""))
                    if not self._owns(*key):
                        continue
                    if self.owned_event_coalesce_s > 0:
                        self.queue.add_after(key, self.owned_event_coalesce_s)
                    else:
                        self.queue.add(key)

    # -- workers -------------------------------------------------------
    def _worker(self) -> None:
        import time as _time
        while not self._stopped.is_set():
            item = self.queue.get(timeout=0.5)
            if item is None:
                continue
            outcome = "success"
            t0 = _time.perf_counter()
            try:
                result = self.reconciler.reconcile(item)
                self.reconcile_count += 1
                self.queue.forget(item)
                if result and result.requeue_after is not None:
                    self.queue.add_after(item, result.requeue_after)
                elif result and result.requeue:
                    self.queue.add_rate_limited(item)
            except ConflictError:
                # optimistic-concurrency loss: immediate-ish retry
                outcome = "conflict"
                self.reconcile_count += 1
                self.queue.add_rate_limited(item)
            except Exception:
                outcome = "error"
                self.error_count += 1
                logger.error("reconcile %s %s failed:\n%s", self.name, item,
                             traceback.format_exc())
                self.queue.add_rate_limited(item)
            finally:
                self.queue.done(item)
                if self.metrics is not None:
                    self.metrics.reconcile_total.labels(self.name, outcome).inc()
                    self.metrics.reconcile_duration.labels(self.name).observe(
                        _time.perf_counter() - t0)
                    self.metrics.workqueue_depth.labels(self.name).set(
                        len(self.queue))

    def start(self) -> None:
        for i in range(self.workers):
            t = threading.Thread(target=self._worker, name=f"{self.name}-{i}", daemon=True)
            t.start()
            self._threads.append(t)

    def stop(self) -> None:
        self._stopped.set()
        self.queue.shutdown()
        for t in self._threads:
            t.join(timeout=2)


class Manager:
    """Hosts controllers over one API server watch stream."""

    def __init__(self, server: InMemoryApiServer):
        self.server = server
        self.controllers: List[Controller] = []
        self._watcher = None
        self._thread: Optional[threading.Thread] = None
        self._stopped = threading.Event()

    def add_controller(self, controller: Controller) -> None:
        self.controllers.append(controller)

    def _dispatch_loop(self) -> None:
        while not self._stopped.is_set():
            ev = self._watcher.next(timeout=0.5)
            if ev is None:
                continue
            event_type, obj = ev
            for c in self.controllers:
                c.observe(event_type, obj)

    def start(self) -> None:
        kinds = set()
        for c in self.controllers:
            kinds.add(c.primary_kind)
            kinds.update(c.owned_kinds)
        self._watcher = self.server.watch(kinds)
        # seed: enqueue existing primaries (informer initial list)
        for c in self.controllers:
            for obj in self.server.list(c.primary_kind):
                c.observe("ADDED", obj)
        self._thread = threading.Thread(target=self._dispatch_loop,
                                        name="manager-dispatch", daemon=True)
        self._thread.start()
        for c in self.controllers:
            c.start()

    def stop(self) -> None:
        self._stopped.set()
        for c in self.controllers:
            c.stop()
        if self._watcher:
            self._watcher.stop()
        if self._thread:
            self._thread.join(timeout=2)

    # -- test helpers --------------------------------------------------
    def wait_idle(self, timeout: float = 10.0) -> bool:
        """Wait until all controller queues drain (best-effort)."""
        import time
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if all(len(c.queue) == 0 for c in self.controllers):
                return True
            time.sleep(0.02)
        return False
