"""Typed Kubernetes client (the client-go / generated-clientset analog).

``KubeClient`` is the single seam reconcilers use; it is implemented by:

* ``InMemoryClient`` — over :class:`~kuberay_amd.kube.store.InMemoryApiServer`
  (tests, bench, local-node runtime),
* ``RestClient`` (kuberay_amd/kube/rest.py) — over a real kube-apiserver via
  httpx, same verb surface, for actual cluster deployments.

Typed objects are pydantic models from ``kuberay_amd.kube.objects`` and
``kuberay_amd.models``; the client converts at the boundary.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Type, TypeVar

from ..models import RayCluster, RayCronJob, RayJob, RayService
from . import objects as k8s
from .store import InMemoryApiServer, NotFoundError

T = TypeVar("T")

KIND_TO_MODEL: Dict[str, Any] = {
    "RayCluster": RayCluster,
    "RayJob": RayJob,
    "RayService": RayService,
    "RayCronJob": RayCronJob,
    "Pod": k8s.Pod,
    "Service": k8s.Service,
    "Secret": k8s.Secret,
    "ConfigMap": k8s.ConfigMap,
    "PersistentVolumeClaim": k8s.PersistentVolumeClaim,
    "ServiceAccount": k8s.ServiceAccount,
    "Role": k8s.Role,
    "RoleBinding": k8s.RoleBinding,
    "Job": k8s.Job,
    "NetworkPolicy": k8s.NetworkPolicy,
    "Ingress": k8s.Ingress,
    "EndpointSlice": k8s.EndpointSlice,
    "Event": k8s.Event,
}


def model_for_kind(kind: str):
    return KIND_TO_MODEL.get(kind)


class KubeClient:
    """Abstract verb surface. All methods accept/return typed models."""

    def create(self, obj: T) -> T:
        raise NotImplementedError

    def get(self, model: Type[T], namespace: str, name: str) -> T:
        raise NotImplementedError

    def try_get(self, model: Type[T], namespace: str, name: str) -> Optional[T]:
        try:
            return self.get(model, namespace, name)
        except NotFoundError:
            return None

    def list(
        self,
        model: Type[T],
        namespace: Optional[str] = None,
        label_selector: Optional[Dict[str, str]] = None,
    ) -> List[T]:
        raise NotImplementedError

    def update(self, obj: T) -> T:
        raise NotImplementedError

    def update_status(self, obj: T) -> T:
        raise NotImplementedError

    def patch(self, model: Type[T], namespace: str, name: str,
              patch: Dict[str, Any], subresource: Optional[str] = None) -> T:
        raise NotImplementedError

    def delete(self, model_or_obj, namespace: Optional[str] = None,
               name: Optional[str] = None) -> None:
        raise NotImplementedError

    def update_with_retry(self, model: Type[T], namespace: str, name: str,
                          mutate, attempts: int = 10,
                          backoff_s: float = 0.02) -> T:
        """Get-mutate-update with optimistic-concurrency retry
        (client-go ``retry.RetryOnConflict`` analog).

        ``mutate(obj)`` is re-applied to a FRESH read on every attempt, so a
        409 from a racing writer (the reconciler, an autoscaler) never
        surfaces to the caller as a failure. All user-facing read-modify-write
        (CLI scale/suspend, python clients, upgrade spec edits) must go
        through this instead of naked ``update``.
        """
        import time as _time

        from .store import ConflictError
        last: Optional[Exception] = None
        for attempt in range(attempts):
            obj = self.get(model, namespace, name)
            mutate(obj)
            try:
                return self.update(obj)
            except ConflictError as e:
                last = e
                _time.sleep(backoff_s * (attempt + 1))
        raise last if last is not None else RuntimeError("unreachable")


class RawObjectClient:
    """Dict-object verbs over either backend: the in-memory server
    directly, or a RestClient's raw_* methods (for kinds with no typed
    model — PodGroup, cert-manager CRs, Gateway API objects, Nodes).

    On the REST backend the kind's path is derived from the object's
    apiVersion when the kind isn't in the static RESOURCES map, so
    schedulers and cert issuance work identically in real deployments.
    """

    def __init__(self, client):
        self.server = getattr(client, "server", None)
        self.client = client

    def try_get(self, kind: str, namespace: str, name: str,
                api_version: Optional[str] = None):
        if self.server is not None:
            return self.server.try_get(kind, namespace, name)
        fn = getattr(self.client, "raw_try_get", None)
        return fn(kind, namespace, name, api_version=api_version) if fn else None

    def create(self, obj: Dict[str, Any]):
        if self.server is not None:
            return self.server.create(obj)
        fn = getattr(self.client, "raw_create", None)
        return fn(obj) if fn else None

    def update(self, obj: Dict[str, Any]):
        if self.server is not None:
            return self.server.update(obj)
        fn = getattr(self.client, "raw_update", None)
        return fn(obj) if fn else None

    def patch(self, kind: str, namespace: str, name: str,
              patch: Dict[str, Any], api_version: Optional[str] = None):
        if self.server is not None:
            return self.server.patch_merge(kind, namespace, name, patch)
        fn = getattr(self.client, "raw_patch", None)
        return fn(kind, namespace, name, patch,
                  api_version=api_version) if fn else None

    def delete(self, kind: str, namespace: str, name: str,
               api_version: Optional[str] = None) -> None:
        if self.server is not None:
            self.server.delete(kind, namespace, name)
            return
        fn = getattr(self.client, "raw_delete", None)
        if fn is not None:
            fn(kind, namespace, name, api_version=api_version)

    def list(self, kind: str, namespace: Optional[str] = None,
             api_version: Optional[str] = None):
        if self.server is not None:
            return self.server.list(kind, namespace)
        fn = getattr(self.client, "raw_list", None)
        return fn(kind, namespace, api_version=api_version) if fn else []


def _kind_of(model_or_obj) -> str:
    if isinstance(model_or_obj, type):
        # model class: read the default of the `kind` field
        return model_or_obj.model_fields["kind"].default
    return model_or_obj.kind


class InMemoryClient(KubeClient):
    def __init__(self, server: Optional[InMemoryApiServer] = None):
        self.server = server or InMemoryApiServer()

    def create(self, obj):
        data = obj.to_dict()
        data["kind"] = obj.kind
        out = self.server.create(data, assume_owned=True)
        return type(obj).from_dict(out)

    def get(self, model, namespace, name):
        out = self.server.get(_kind_of(model), namespace, name)
        return model.from_dict(out)

    def list(self, model, namespace=None, label_selector=None):
        kind = _kind_of(model)
        return [model.from_dict(o) for o in self.server.list(kind, namespace, label_selector)]

    def update(self, obj):
        out = self.server.update(obj.to_dict(), assume_owned=True)
        # client-go semantics: refresh the passed object's resourceVersion
        obj.metadata.resource_version = out["metadata"]["resourceVersion"]
        obj.metadata.generation = out["metadata"].get("generation")
        return type(obj).from_dict(out)

    def update_status(self, obj):
        out = self.server.update(obj.to_dict(), subresource="status",
                                 assume_owned=True)
        obj.metadata.resource_version = out["metadata"]["resourceVersion"]
        return type(obj).from_dict(out)

    def patch(self, model, namespace, name, patch, subresource=None):
        out = self.server.patch_merge(_kind_of(model), namespace, name, patch,
                                      subresource=subresource)
        return model.from_dict(out)

    def list_pod_views(self, namespace=None, label_selector=None):
        """Reconcile hot-loop projection (no pod materialization)."""
        return self.server.list_pod_views(namespace, label_selector)

    def delete(self, model_or_obj, namespace=None, name=None):
        if namespace is None:
            namespace = model_or_obj.metadata.namespace or "default"
            name = model_or_obj.metadata.name
        self.server.delete(_kind_of(model_or_obj), namespace, name)
