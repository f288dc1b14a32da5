"""NativeBackend — the storage Backend over the C++ engine
(kuberay_amd/_native/engine.cpp).

Objects are serialized once per write (compact JSON, CPython's C encoder)
and live in the C++ heap; reads parse on demand; the reconcile hot loop
reads precomputed pod views without touching the blob at all.
"""
from __future__ import annotations

import json
from typing import Any, Dict, List, Optional

from .._native import engine  # raises ImportError if not built
from .store import Key, PodView

_dumps = json.dumps
_loads = json.loads


def _view_of(obj: Dict[str, Any]):
    status = obj.get("status") or {}
    spec = obj.get("spec") or {}
    meta = obj.get("metadata", {})
    v = engine.PodViewData()
    v.name = meta.get("name", "")
    v.ns = meta.get("namespace", "default")
    v.phase = status.get("phase") or ""
    v.deletion_ts = meta.get("deletionTimestamp") or ""
    v.pod_ip = status.get("podIP") or ""
    v.restart_policy = spec.get("restartPolicy") or ""
    v.creation_ts = meta.get("creationTimestamp") or ""
    v.ready = any(c.get("type") == "Ready" and c.get("status") == "True"
                  for c in status.get("conditions") or [])
    v.terminated = False
    containers = spec.get("containers") or []
    if containers:
        ray_name = containers[0].get("name")
        for cs in status.get("containerStatuses") or []:
            if cs.get("name") == ray_name and (cs.get("state") or {}).get("terminated"):
                v.terminated = True
                break
    return v


class NativeBackend:
    name = "native-cpp"

    def __init__(self) -> None:
        self._store = engine.NativeStore()
        # kind -> live-object count; NativeStore has no kind enumeration,
        # and the HTTP facade needs one for dynamic (CRD) path routing
        self._kind_counts: Dict[str, int] = {}

    def put(self, key: Key, obj: Dict[str, Any]) -> None:
        kind, ns, name = key
        if not self._store.contains(kind, ns, name):
            self._kind_counts[kind] = self._kind_counts.get(kind, 0) + 1
        meta = obj.get("metadata", {})
        labels = list((meta.get("labels") or {}).items())
        owners = [ref.get("uid") for ref in meta.get("ownerReferences") or []
                  if ref.get("uid")]
        view = _view_of(obj) if kind == "Pod" else None
        blob = _dumps(obj, separators=(",", ":"))
        self._store.put(kind, ns, name, blob, meta.get("resourceVersion", ""),
                        labels, owners, view)

    def fetch(self, key: Key) -> Optional[Dict[str, Any]]:
        blob = self._store.fetch(*key)
        return _loads(blob) if blob is not None else None

    def rv(self, key: Key) -> Optional[str]:
        return self._store.rv(*key)

    def contains(self, key: Key) -> bool:
        return self._store.contains(*key)

    def remove(self, key: Key) -> Optional[Dict[str, Any]]:
        blob = self._store.remove(*key)
        if blob is not None:
            n = self._kind_counts.get(key[0], 0) - 1
            if n > 0:
                self._kind_counts[key[0]] = n
            else:
                self._kind_counts.pop(key[0], None)
        return _loads(blob) if blob is not None else None

    def list(self, kind: str, namespace: Optional[str],
             selector: Optional[Dict[str, str]]) -> List[Dict[str, Any]]:
        blobs = self._store.list_blobs(kind, namespace,
                                       list((selector or {}).items()))
        return [_loads(b) for b in blobs]

    def list_views(self, namespace: Optional[str],
                   selector: Optional[Dict[str, str]]) -> List[PodView]:
        rows = self._store.list_views(namespace, list((selector or {}).items()))
        return [
            PodView(name=r[0], namespace=r[1], labels=r[2], phase=r[3],
                    ready=r[4], deletion_timestamp=r[5], pod_ip=r[6],
                    restart_policy=r[7], ray_container_terminated=r[8],
                    creation_timestamp=r[9])
            for r in rows
        ]

    def dependents(self, uid: str) -> List[Key]:
        return [tuple(t) for t in self._store.dependents(uid)]

    def drop_owner(self, uid: str) -> None:
        self._store.drop_owner(uid)

    def count(self, kind: str) -> int:
        return self._store.count(kind)

    def total_bytes(self) -> int:
        return self._store.total_bytes()

    def kinds(self) -> List[str]:
        return list(self._kind_counts)
