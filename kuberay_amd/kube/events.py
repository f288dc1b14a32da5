"""K8s Event recording (record.EventRecorder analog).

The reference emits an Event on every mutation (Recorder.Eventf throughout
the reconcilers). Events are aggregated by (object, reason, message) with a
count, like the real apiserver does, to keep the store small at 500-cluster
scale.
"""
from __future__ import annotations

import threading
from typing import Any

from .store import InMemoryApiServer, NotFoundError, now_iso


class EventRecorder:
    def eventf(self, obj: Any, event_type: str, reason: str, message: str, *args) -> None:
        raise NotImplementedError


class NullRecorder(EventRecorder):
    def eventf(self, obj, event_type, reason, message, *args) -> None:
        pass


class StoreRecorder(EventRecorder):
    def __init__(self, server: InMemoryApiServer):
        self.server = server
        self._lock = threading.Lock()

    def eventf(self, obj, event_type, reason, message, *args) -> None:
        if args:
            message = message % args
        if hasattr(obj, "metadata"):
            namespace = obj.metadata.namespace or "default"
            name = obj.metadata.name
            kind = obj.kind
            uid = obj.metadata.uid
        else:
            meta = obj.get("metadata", {})
            namespace, name = meta.get("namespace", "default"), meta.get("name")
            kind, uid = obj.get("kind"), meta.get("uid")
        ev_name = f"{name}.{abs(hash((kind, reason, message))) % 10**10:x}"
        with self._lock:
            existing = self.server.try_get("Event", namespace, ev_name)
            if existing is not None:
                try:
                    self.server.patch_merge(
                        "Event", namespace, ev_name,
                        {"count": (existing.get("count") or 1) + 1,
                         "lastTimestamp": now_iso()})
                    return
                except NotFoundError:
                    pass
            self.server.create({
                "apiVersion": "v1",
                "kind": "Event",
                "metadata": {"name": ev_name, "namespace": namespace},
                "type": event_type,
                "reason": reason,
                "message": message,
                "involvedObject": {"kind": kind, "namespace": namespace,
                                   "name": name, "uid": uid},
                "count": 1,
                "firstTimestamp": now_iso(),
                "lastTimestamp": now_iso(),
            })
