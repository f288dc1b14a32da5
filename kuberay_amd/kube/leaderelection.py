"""Lease-based leader election (controller-runtime analog; reference:
main.go leader-election flags).

One operator replica holds a ``coordination.k8s.io/v1`` Lease; the others
stand by and take over when renewals stop. Works over the in-memory server
(dict Lease objects) or a RestClient's raw verbs on a real cluster.
"""
from __future__ import annotations

import threading
import time
import uuid
from typing import Callable, Optional

from .store import AlreadyExistsError, ConflictError, NotFoundError

LEASE_KIND = "Lease"


def _now_micro() -> str:
    t = time.time()
    frac = int((t % 1) * 1e6)
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(t)) + f".{frac:06d}Z"


def _parse_micro(ts: str) -> float:
    import calendar
    try:
        base, _, rest = ts.partition(".")
        whole = calendar.timegm(time.strptime(base, "%Y-%m-%dT%H:%M:%S"))
        frac = float("0." + rest.rstrip("Z")) if rest else 0.0
        return whole + frac
    except (ValueError, AttributeError):
        return 0.0


class _LeaseStore:
    """Raw Lease verbs over either backend."""

    def __init__(self, client):
        self.server = getattr(client, "server", None)
        self.client = client

    def get(self, namespace, name):
        if self.server is not None:
            return self.server.try_get(LEASE_KIND, namespace, name)
        fn = getattr(self.client, "raw_try_get", None)
        return fn(LEASE_KIND, namespace, name) if fn else None

    def create(self, obj):
        if self.server is not None:
            return self.server.create(obj)
        return self.client.raw_create(obj)

    def update(self, obj):
        """Full-object PUT carrying the resourceVersion read by the caller.

        Both backends enforce the optimistic-concurrency precondition and
        raise ConflictError on a stale write, so two standbys racing to take
        over an expired lease cannot both win (client-go leaderelection
        semantics; a merge patch here would allow split-brain).
        """
        if self.server is not None:
            return self.server.update(obj)
        fn = getattr(self.client, "raw_update", None)
        if fn is not None:
            return fn(obj)
        raise ConflictError("lease backend lacks raw_update; refusing "
                            "non-atomic lease takeover")


class LeaderElector:
    def __init__(self, client, lease_name: str = "kuberay-amd-operator",
                 namespace: str = "ray-system",
                 identity: Optional[str] = None,
                 lease_duration_s: float = 15.0,
                 renew_period_s: float = 5.0,
                 on_started_leading: Optional[Callable[[], None]] = None,
                 on_stopped_leading: Optional[Callable[[], None]] = None):
        self.store = _LeaseStore(client)
        self.lease_name = lease_name
        self.namespace = namespace
        self.identity = identity or f"{uuid.uuid4().hex[:8]}"
        self.lease_duration_s = lease_duration_s
        self.renew_period_s = renew_period_s
        self.on_started_leading = on_started_leading
        self.on_stopped_leading = on_stopped_leading
        self.is_leader = False
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------------
    def try_acquire_or_renew(self) -> bool:
        """One election round. Returns current leadership."""
        lease = self.store.get(self.namespace, self.lease_name)
        if lease is None:
            try:
                self.store.create({
                    "apiVersion": "coordination.k8s.io/v1",
                    "kind": LEASE_KIND,
                    "metadata": {"name": self.lease_name,
                                 "namespace": self.namespace},
                    "spec": self._spec(acquisitions=1),
                })
                return self._become(True)
            except AlreadyExistsError:
                return self._become(False)

        spec = lease.get("spec", {})
        holder = spec.get("holderIdentity")
        renew = _parse_micro(spec.get("renewTime", ""))
        duration = spec.get("leaseDurationSeconds", self.lease_duration_s)
        expired = (time.time() - renew) > duration

        if holder == self.identity or expired or not holder:
            lease["spec"] = self._spec(
                acquisitions=(spec.get("leaseTransitions", 0) or 0)
                + (0 if holder == self.identity else 1))
            try:
                self.store.update(lease)
                return self._become(True)
            except (ConflictError, NotFoundError):
                return self._become(False)
        return self._become(False)

    def _spec(self, acquisitions: int) -> dict:
        return {
            "holderIdentity": self.identity,
            # int on the wire for real K8s; sub-second floats kept for tests
            "leaseDurationSeconds": (int(self.lease_duration_s)
                                     if self.lease_duration_s >= 1
                                     else self.lease_duration_s),
            "renewTime": _now_micro(),
            "acquireTime": _now_micro(),
            "leaseTransitions": acquisitions,
        }

    def _become(self, leader: bool) -> bool:
        if leader and not self.is_leader:
            self.is_leader = True
            if self.on_started_leading:
                self.on_started_leading()
        elif not leader and self.is_leader:
            self.is_leader = False
            if self.on_stopped_leading:
                self.on_stopped_leading()
        return self.is_leader

    # ------------------------------------------------------------------
    def run(self) -> None:
        while not self._stop.is_set():
            try:
                self.try_acquire_or_renew()
            except Exception:
                pass
            self._stop.wait(self.renew_period_s)
        # graceful release
        if self.is_leader:
            lease = self.store.get(self.namespace, self.lease_name)
            if lease is not None and lease.get("spec", {}).get(
                    "holderIdentity") == self.identity:
                lease["spec"]["holderIdentity"] = ""
                try:
                    self.store.update(lease)
                except (ConflictError, NotFoundError):
                    pass
            self._become(False)

    def start(self) -> None:
        self._thread = threading.Thread(target=self.run, name="leader-election",
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=3)
