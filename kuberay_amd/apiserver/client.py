"""Typed Python client for the v1 APIServer
(reference: clients/python-apiserver-client — a requests-based typed client
for the simplified api.Cluster/ComputeTemplate/RayJob/RayService surface).

Speaks the same JSON shapes as the HTTP v1 endpoints (`apiserver/app.py`)
and kray's server mode; works against any base URL (or an injected httpx
client in tests).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

import httpx


class ApiServerError(Exception):
    def __init__(self, status_code: int, message: str):
        super().__init__(f"{status_code}: {message}")
        self.status_code = status_code


class ApiServerClient:
    def __init__(self, base_url: str = "http://127.0.0.1:8888",
                 token: Optional[str] = None,
                 http_client: Optional[httpx.Client] = None,
                 timeout: float = 30.0):
        if http_client is not None:
            self._http = http_client
        else:
            headers = {"Authorization": f"Bearer {token}"} if token else {}
            self._http = httpx.Client(base_url=base_url, timeout=timeout,
                                      headers=headers)

    # -- plumbing -------------------------------------------------------
    def _url(self, namespace: str, resource: str,
             name: Optional[str] = None) -> str:
        p = f"/apis/v1/namespaces/{namespace}/{resource}"
        return f"{p}/{name}" if name else p

    def _check(self, resp: httpx.Response) -> Dict[str, Any]:
        if resp.status_code >= 400:
            raise ApiServerError(resp.status_code, resp.text[:300])
        return resp.json() if resp.content else {}

    # -- compute templates ---------------------------------------------
    def create_compute_template(self, namespace: str,
                                template: Dict[str, Any]) -> Dict[str, Any]:
        return self._check(self._http.post(
            self._url(namespace, "compute_templates"), json=template))

    def get_compute_template(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._check(self._http.get(
            self._url(namespace, "compute_templates", name)))

    def list_compute_templates(self, namespace: str) -> List[Dict[str, Any]]:
        return self._check(self._http.get(
            self._url(namespace, "compute_templates"))).get(
            "computeTemplates", [])

    def delete_compute_template(self, namespace: str, name: str) -> None:
        self._check(self._http.delete(
            self._url(namespace, "compute_templates", name)))

    # -- image templates (ImageTemplateService HTTP mapping) ------------
    def create_image_template(self, namespace: str,
                              template: Dict[str, Any]) -> Dict[str, Any]:
        return self._check(self._http.post(
            self._url(namespace, "image_templates"), json=template))

    def get_image_template(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._check(self._http.get(
            self._url(namespace, "image_templates", name)))

    def list_image_templates(self, namespace: str) -> List[Dict[str, Any]]:
        return self._check(self._http.get(
            self._url(namespace, "image_templates"))).get(
            "imageTemplates", [])

    def delete_image_template(self, namespace: str, name: str) -> None:
        self._check(self._http.delete(
            self._url(namespace, "image_templates", name)))

    # -- clusters -------------------------------------------------------
    def create_cluster(self, namespace: str,
                       cluster: Dict[str, Any]) -> Dict[str, Any]:
        return self._check(self._http.post(
            self._url(namespace, "clusters"), json=cluster))

    def get_cluster(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._check(self._http.get(
            self._url(namespace, "clusters", name)))

    def list_clusters(self, namespace: str) -> List[Dict[str, Any]]:
        return self._check(self._http.get(
            self._url(namespace, "clusters"))).get("clusters", [])

    def delete_cluster(self, namespace: str, name: str) -> None:
        self._check(self._http.delete(self._url(namespace, "clusters", name)))

    def wait_until_cluster_running(self, namespace: str, name: str,
                                   timeout_s: float = 300.0,
                                   poll_s: float = 1.0) -> bool:
        import time
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline:
            if self.get_cluster(namespace, name).get("clusterState") == "ready":
                return True
            time.sleep(poll_s)
        return False

    # -- jobs -----------------------------------------------------------
    def create_job(self, namespace: str, job: Dict[str, Any]) -> Dict[str, Any]:
        return self._check(self._http.post(
            self._url(namespace, "jobs"), json=job))

    def get_job(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._check(self._http.get(self._url(namespace, "jobs", name)))

    def list_jobs(self, namespace: str) -> List[Dict[str, Any]]:
        return self._check(self._http.get(
            self._url(namespace, "jobs"))).get("jobs", [])

    def delete_job(self, namespace: str, name: str) -> None:
        self._check(self._http.delete(self._url(namespace, "jobs", name)))

    # -- services -------------------------------------------------------
    def create_service(self, namespace: str,
                       service: Dict[str, Any]) -> Dict[str, Any]:
        return self._check(self._http.post(
            self._url(namespace, "services"), json=service))

    def get_service(self, namespace: str, name: str) -> Dict[str, Any]:
        return self._check(self._http.get(
            self._url(namespace, "services", name)))

    def list_services(self, namespace: str) -> List[Dict[str, Any]]:
        return self._check(self._http.get(
            self._url(namespace, "services"))).get("services", [])

    def delete_service(self, namespace: str, name: str) -> None:
        self._check(self._http.delete(self._url(namespace, "services", name)))

