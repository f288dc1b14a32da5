"""In-memory Kubernetes API server (the envtest analog, plus a GC).

The reference tests its reconcilers against a real kube-apiserver spun up by
controller-runtime's envtest (ray-operator/controllers/ray/suite_test.go:77-120)
with NO kubelet — pods never actually run. This module provides the same
seam natively: a thread-safe object store with resourceVersion semantics,
optimistic-concurrency conflicts, finalizer-aware deletion, ownerReference
garbage collection and label-selector lists + watches.

It is also the substrate of the scale benchmark (500-cluster
soak): all verbs are O(1) dict ops except list, which uses per-(kind,label)
indices. A C++ native backend with the same verb surface lives in
``kuberay_amd._native`` and is used when built.
"""
from __future__ import annotations

import copy
import fnmatch
import threading
import time
import uuid
from collections import defaultdict
from typing import Any, Callable, Dict, Iterable, List, Optional, Tuple

Key = Tuple[str, str, str]  # (kind, namespace, name)


def jsoncopy(obj):
    """Deep copy for JSON-shaped trees (dict/list/scalars only) — ~4x faster
    than copy.deepcopy, which pays for memoization and type dispatch these
    trees never need."""
    t = type(obj)
    if t is dict:
        return {k: jsoncopy(v) for k, v in obj.items()}
    if t is list:
        return [jsoncopy(v) for v in obj]
    return obj


class ApiError(Exception):
    def __init__(self, code: int, message: str):
        super().__init__(f"{code}: {message}")
        self.code = code
        self.message = message


class NotFoundError(ApiError):
    def __init__(self, message: str = "not found"):
        super().__init__(404, message)


class ConflictError(ApiError):
    def __init__(self, message: str = "conflict"):
        super().__init__(409, message)


class AlreadyExistsError(ApiError):
    def __init__(self, message: str = "already exists"):
        super().__init__(409, message)


def now_iso() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


def match_labels(labels: Optional[Dict[str, str]], selector: Optional[Dict[str, str]]) -> bool:
    if not selector:
        return True
    labels = labels or {}
    return all(labels.get(k) == v for k, v in selector.items())


class InMemoryApiServer:
    """Thread-safe object store with Kubernetes verb semantics."""

    def __init__(self) -> None:
        self._lock = threading.RLock()
        self._objects: Dict[Key, Dict[str, Any]] = {}
        self._rv = 0
        self._watchers: List["Watcher"] = []
        # label index: (kind, label_key, label_value) -> set of keys
        self._label_index: Dict[Tuple[str, str, str], set] = defaultdict(set)
        self._kind_index: Dict[str, set] = defaultdict(set)
        self._owner_index: Dict[str, set] = defaultdict(set)  # owner uid -> keys

    # -- internals -----------------------------------------------------
    def _next_rv(self) -> str:
        self._rv += 1
        return str(self._rv)

    def _index_add(self, key: Key, obj: Dict[str, Any]) -> None:
        self._kind_index[key[0]].add(key)
        meta = obj.get("metadata", {})
        for lk, lv in (meta.get("labels") or {}).items():
            self._label_index[(key[0], lk, lv)].add(key)
        for ref in meta.get("ownerReferences") or []:
            uid = ref.get("uid")
            if uid:
                self._owner_index[uid].add(key)

    def _index_remove(self, key: Key, obj: Dict[str, Any]) -> None:
        self._kind_index[key[0]].discard(key)
        meta = obj.get("metadata", {})
        for lk, lv in (meta.get("labels") or {}).items():
            self._label_index[(key[0], lk, lv)].discard(key)
        for ref in meta.get("ownerReferences") or []:
            uid = ref.get("uid")
            if uid:
                self._owner_index[uid].discard(key)

    def _notify(self, event_type: str, obj: Dict[str, Any]) -> None:
        for w in list(self._watchers):
            w.push(event_type, obj)

    @staticmethod
    def _key_of(obj: Dict[str, Any]) -> Key:
        meta = obj.get("metadata", {})
        return (obj.get("kind", ""), meta.get("namespace", "default"), meta.get("name", ""))

    # -- verbs ---------------------------------------------------------
    def create(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        obj = jsoncopy(obj)
        meta = obj.setdefault("metadata", {})
        meta.setdefault("namespace", "default")
        if not meta.get("name"):
            gen = meta.get("generateName")
            if not gen:
                raise ApiError(422, "name or generateName required")
            meta["name"] = gen + uuid.uuid4().hex[:5]
        with self._lock:
            key = self._key_of(obj)
            if key in self._objects:
                raise AlreadyExistsError(f"{key} already exists")
            meta["uid"] = str(uuid.uuid4())
            meta["resourceVersion"] = self._next_rv()
            meta["generation"] = 1
            meta["creationTimestamp"] = now_iso()
            self._objects[key] = obj
            self._index_add(key, obj)
            out = jsoncopy(obj)
        self._notify("ADDED", out)
        return out

    def get(self, kind: str, namespace: str, name: str) -> Dict[str, Any]:
        with self._lock:
            obj = self._objects.get((kind, namespace, name))
            if obj is None:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            return jsoncopy(obj)

    def try_get(self, kind: str, namespace: str, name: str) -> Optional[Dict[str, Any]]:
        try:
            return self.get(kind, namespace, name)
        except NotFoundError:
            return None

    def list(
        self,
        kind: str,
        namespace: Optional[str] = None,
        label_selector: Optional[Dict[str, str]] = None,
    ) -> List[Dict[str, Any]]:
        with self._lock:
            if label_selector:
                # use the most selective label index entry
                candidate_sets = [
                    self._label_index.get((kind, k, v), set())
                    for k, v in label_selector.items()
                ]
                keys = set.intersection(*candidate_sets) if candidate_sets else set()
            else:
                keys = set(self._kind_index.get(kind, set()))
            out = []
            for key in keys:
                if namespace is not None and key[1] != namespace:
                    continue
                obj = self._objects.get(key)
                if obj is None:
                    continue
                if match_labels(obj.get("metadata", {}).get("labels"), label_selector):
                    out.append(jsoncopy(obj))
            out.sort(key=lambda o: (o["metadata"]["namespace"], o["metadata"]["name"]))
            return out

    def update(self, obj: Dict[str, Any], *, subresource: Optional[str] = None) -> Dict[str, Any]:
        obj = jsoncopy(obj)
        key = self._key_of(obj)
        with self._lock:
            current = self._objects.get(key)
            if current is None:
                raise NotFoundError(f"{key} not found")
            meta = obj.setdefault("metadata", {})
            rv = meta.get("resourceVersion")
            if rv is not None and rv != current["metadata"]["resourceVersion"]:
                raise ConflictError(
                    f"{key}: resourceVersion mismatch {rv} != {current['metadata']['resourceVersion']}"
                )
            self._index_remove(key, current)
            if subresource == "status":
                # status updates only replace .status
                new_obj = jsoncopy(current)
                new_obj["status"] = obj.get("status", {})
            else:
                new_obj = obj
                # spec changes bump generation
                if new_obj.get("spec") != current.get("spec"):
                    new_obj["metadata"]["generation"] = current["metadata"].get("generation", 1) + 1
                else:
                    new_obj["metadata"]["generation"] = current["metadata"].get("generation", 1)
                # status only changes through the status subresource
                new_obj["status"] = current.get("status", {})
            new_obj["metadata"]["uid"] = current["metadata"]["uid"]
            new_obj["metadata"]["creationTimestamp"] = current["metadata"]["creationTimestamp"]
            if current["metadata"].get("deletionTimestamp"):
                new_obj["metadata"]["deletionTimestamp"] = current["metadata"]["deletionTimestamp"]
            new_obj["metadata"]["resourceVersion"] = self._next_rv()
            self._objects[key] = new_obj
            self._index_add(key, new_obj)
            finalizers_gone = (
                current["metadata"].get("deletionTimestamp")
                and not new_obj["metadata"].get("finalizers")
            )
            out = jsoncopy(new_obj)
        self._notify("MODIFIED", out)
        if finalizers_gone:
            # terminating object dropped its last finalizer -> actually delete
            self._finalize_delete(key)
        return out

    def patch_merge(
        self, kind: str, namespace: str, name: str, patch: Dict[str, Any],
        *, subresource: Optional[str] = None,
    ) -> Dict[str, Any]:
        """Strategic-merge-ish patch (recursive dict merge; lists replaced)."""
        with self._lock:
            current = self._objects.get((kind, namespace, name))
            if current is None:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            merged = jsoncopy(current)

            def merge(dst, src):
                for k, v in src.items():
                    if isinstance(v, dict) and isinstance(dst.get(k), dict):
                        merge(dst[k], v)
                    elif v is None:
                        dst.pop(k, None)
                    else:
                        dst[k] = jsoncopy(v)

            merge(merged, patch)
            merged["metadata"]["resourceVersion"] = current["metadata"]["resourceVersion"]
        return self.update(merged, subresource=subresource)

    def delete(self, kind: str, namespace: str, name: str) -> None:
        with self._lock:
            key = (kind, namespace, name)
            current = self._objects.get(key)
            if current is None:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            if current["metadata"].get("finalizers"):
                if not current["metadata"].get("deletionTimestamp"):
                    self._index_remove(key, current)
                    current["metadata"]["deletionTimestamp"] = now_iso()
                    current["metadata"]["resourceVersion"] = self._next_rv()
                    self._index_add(key, current)
                    out = copy.deepcopy(current)
                else:
                    return
            else:
                out = None
        if out is not None:
            self._notify("MODIFIED", out)
            return
        self._finalize_delete(key)

    def _finalize_delete(self, key: Key) -> None:
        with self._lock:
            current = self._objects.pop(key, None)
            if current is None:
                return
            self._index_remove(key, current)
            uid = current["metadata"]["uid"]
            # ownerReference GC via the owner-uid index
            dependents = list(self._owner_index.pop(uid, ()))
        self._notify("DELETED", current)
        for dep in dependents:
            try:
                self.delete(*dep)
            except NotFoundError:
                pass

    # -- watches -------------------------------------------------------
    def watch(self, kinds: Optional[Iterable[str]] = None) -> "Watcher":
        w = Watcher(self, set(kinds) if kinds else None)
        with self._lock:
            self._watchers.append(w)
        return w

    def stop_watch(self, w: "Watcher") -> None:
        with self._lock:
            if w in self._watchers:
                self._watchers.remove(w)

    # -- introspection -------------------------------------------------
    def count(self, kind: str) -> int:
        with self._lock:
            return len(self._kind_index.get(kind, set()))


class Watcher:
    """A watch stream: buffered (event_type, object) pairs."""

    def __init__(self, server: InMemoryApiServer, kinds: Optional[set]):
        self._server = server
        self._kinds = kinds
        self._cond = threading.Condition()
        self._events: List[Tuple[str, Dict[str, Any]]] = []
        self._stopped = False

    def push(self, event_type: str, obj: Dict[str, Any]) -> None:
        if self._kinds is not None and obj.get("kind") not in self._kinds:
            return
        with self._cond:
            if self._stopped:
                return
            self._events.append((event_type, obj))
            self._cond.notify_all()

    def next(self, timeout: Optional[float] = None) -> Optional[Tuple[str, Dict[str, Any]]]:
        with self._cond:
            if not self._events:
                self._cond.wait(timeout)
            if self._events:
                return self._events.pop(0)
            return None

    def drain(self) -> List[Tuple[str, Dict[str, Any]]]:
        with self._cond:
            events, self._events = self._events, []
            return events

    def stop(self) -> None:
        with self._cond:
            self._stopped = True
            self._cond.notify_all()
        self._server.stop_watch(self)
