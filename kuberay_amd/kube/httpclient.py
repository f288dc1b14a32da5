"""KubeClient over the v2 (ray.io-restricted) HTTP proxy.

Gives the CLI and external tools the same typed verb surface as the
in-memory client, through kuberay_amd.apiserver's v2 routes (or any real
K8s apiserver exposing /apis/ray.io/v1)."""
from __future__ import annotations

from typing import Optional

import httpx

from .client import KubeClient, _kind_of
from .store import AlreadyExistsError, ApiError, NotFoundError

PLURALS = {"RayCluster": "rayclusters", "RayJob": "rayjobs",
           "RayService": "rayservices", "RayCronJob": "raycronjobs"}


class HttpKubeClient(KubeClient):
    def __init__(self, base_url: str, token: Optional[str] = None,
                 transport=None, timeout: float = 10.0, http_client=None):
        if http_client is not None:
            self._http = http_client
            return
        headers = {}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._http = httpx.Client(base_url=base_url.rstrip("/"),
                                  headers=headers, timeout=timeout,
                                  transport=transport)

    def _path(self, kind: str, namespace: str, name: Optional[str] = None) -> str:
        plural = PLURALS.get(kind)
        if plural is None:
            raise ApiError(400, f"kind {kind} not supported over HTTP client")
        p = f"/apis/ray.io/v1/namespaces/{namespace}/{plural}"
        return f"{p}/{name}" if name else p

    def _request_with_retry(self, method: str, path: str, **kwargs):
        """Retry with backoff on transient transport errors / 5xx
        (apiserversdk proxy.go:106-208 retry round-tripper analog)."""
        import time
        last = None
        for attempt in range(3):
            try:
                resp = self._http.request(method, path, **kwargs)
            except httpx.TransportError as e:
                last = e
                time.sleep(0.1 * (2 ** attempt))
                continue
            if resp.status_code >= 500:
                last = ApiError(resp.status_code, resp.text[:200])
                time.sleep(0.1 * (2 ** attempt))
                continue
            return resp
        if isinstance(last, ApiError):
            raise last
        raise ApiError(503, f"transport error after retries: {last}")

    @staticmethod
    def _check(resp: httpx.Response):
        if resp.status_code == 404:
            raise NotFoundError(resp.text[:200])
        if resp.status_code == 409:
            raise AlreadyExistsError(resp.text[:200])
        if resp.status_code >= 400:
            raise ApiError(resp.status_code, resp.text[:500])
        return resp

    def create(self, obj):
        kind = obj.kind
        ns = obj.metadata.namespace or "default"
        resp = self._check(self._request_with_retry("POST", self._path(kind, ns), json=obj.to_dict()))
        return type(obj).from_dict(resp.json())

    def get(self, model, namespace, name):
        kind = _kind_of(model)
        resp = self._check(self._request_with_retry("GET", self._path(kind, namespace, name)))
        return model.from_dict(resp.json())

    def list(self, model, namespace=None, label_selector=None):
        kind = _kind_of(model)
        ns = namespace or "default"
        resp = self._check(self._request_with_retry("GET", self._path(kind, ns)))
        items = [model.from_dict(o) for o in resp.json().get("items", [])]
        if label_selector:
            items = [o for o in items
                     if all((o.metadata.labels or {}).get(k) == v
                            for k, v in label_selector.items())]
        return items

    def update(self, obj):
        kind = obj.kind
        ns = obj.metadata.namespace or "default"
        resp = self._check(self._request_with_retry("PUT",
            self._path(kind, ns, obj.metadata.name), json=obj.to_dict()))
        out = resp.json()
        obj.metadata.resource_version = out.get("metadata", {}).get("resourceVersion")
        return type(obj).from_dict(out)

    def update_status(self, obj):
        kind = obj.kind
        ns = obj.metadata.namespace or "default"
        resp = self._check(self._request_with_retry("PUT", 
            self._path(kind, ns, obj.metadata.name) + "/status",
            json=obj.to_dict()))
        return type(obj).from_dict(resp.json())

    def patch(self, model, namespace, name, patch, subresource=None):
        kind = _kind_of(model)
        path = self._path(kind, namespace, name)
        if subresource:
            path += f"/{subresource}"
        resp = self._check(self._request_with_retry("PATCH", path, json=patch))
        return model.from_dict(resp.json())

    def delete(self, model_or_obj, namespace=None, name=None):
        if namespace is None:
            namespace = model_or_obj.metadata.namespace or "default"
            name = model_or_obj.metadata.name
        kind = _kind_of(model_or_obj)
        self._check(self._request_with_retry("DELETE", self._path(kind, namespace, name)))
