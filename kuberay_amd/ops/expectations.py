"""Scale expectations — informer-staleness guard
(reference: ray-operator/controllers/ray/expectations/scale_expectations.go).

After the reconciler creates or deletes a pod, the (cluster, group) scale is
"unsatisfied" until the cache reflects the mutation (or a timeout passes),
preventing double-create/double-delete storms when the informer lags the API
server. With the in-memory API server there is no lag, but the guard is kept
for the REST/informer backend and exercised by unit tests.
"""
from __future__ import annotations

import threading
import time
from typing import Dict, Tuple

GroupKey = Tuple[str, str, str]  # (namespace, cluster, group)

EXPECTATION_TIMEOUT_S = 30.0

HEAD_GROUP = "__head__"


class ScaleExpectations:
    """Interface (scale_expectations.go:37-41)."""

    def expect_create_pod(self, namespace: str, cluster: str, group: str, pod_name: str) -> None:
        raise NotImplementedError

    def expect_delete_pod(self, namespace: str, cluster: str, group: str, pod_name: str) -> None:
        raise NotImplementedError

    def is_satisfied(self, cache, namespace: str, cluster: str, group: str) -> bool:
        raise NotImplementedError

    def delete(self, namespace: str, cluster: str) -> None:
        raise NotImplementedError


class RayClusterScaleExpectations(ScaleExpectations):
    def __init__(self, timeout_s: float = EXPECTATION_TIMEOUT_S):
        self._lock = threading.Lock()
        # key -> {pod_name: (op, deadline)}  op in {"create","delete"}
        self._pending: Dict[GroupKey, Dict[str, Tuple[str, float]]] = {}
        self.timeout_s = timeout_s

    def _expect(self, key: GroupKey, pod_name: str, op: str) -> None:
        with self._lock:
            self._pending.setdefault(key, {})[pod_name] = (op, time.monotonic() + self.timeout_s)

    def expect_create_pod(self, namespace, cluster, group, pod_name) -> None:
        self._expect((namespace, cluster, group), pod_name, "create")

    def expect_delete_pod(self, namespace, cluster, group, pod_name) -> None:
        self._expect((namespace, cluster, group), pod_name, "delete")

    def is_satisfied(self, cache, namespace, cluster, group) -> bool:
        """cache: InMemoryApiServer-shaped (contains()/try_get())."""
        key = (namespace, cluster, group)
        now = time.monotonic()
        contains = getattr(cache, "contains", None)
        with self._lock:
            pending = self._pending.get(key)
            if not pending:
                return True
            satisfied = []
            for pod_name, (op, deadline) in pending.items():
                if op == "create":
                    if contains is not None:
                        ok = contains("Pod", namespace, pod_name)
                    else:
                        ok = cache.try_get("Pod", namespace, pod_name) is not None
                else:
                    if contains is not None and not contains("Pod", namespace, pod_name):
                        ok = True
                    else:
                        observed = cache.try_get("Pod", namespace, pod_name)
                        ok = observed is None or bool(
                            observed.get("metadata", {}).get("deletionTimestamp"))
                if ok or now > deadline:
                    satisfied.append(pod_name)
            for pod_name in satisfied:
                pending.pop(pod_name, None)
            if not pending:
                self._pending.pop(key, None)
                return True
            return False

    def delete(self, namespace, cluster) -> None:
        with self._lock:
            for key in [k for k in self._pending if k[0] == namespace and k[1] == cluster]:
                self._pending.pop(key, None)


class FakeScaleExpectations(ScaleExpectations):
    """Always satisfied (scale_expectations.go:166-179 analog)."""

    def expect_create_pod(self, *a) -> None:
        pass

    def expect_delete_pod(self, *a) -> None:
        pass

    def is_satisfied(self, *a) -> bool:
        return True

    def delete(self, *a) -> None:
        pass
