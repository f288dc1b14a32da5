"""Ray dashboard HTTP client
(reference: utils/dashboardclient/dashboard_httpclient.go:29-41).

The only data-plane touch point of the operator: job submission state and
Serve application state are read/written through the head pod's dashboard
(:8265). ``RayDashboardClient`` talks real HTTP via httpx; tests and the
in-process harness inject ``FakeRayDashboardClient``
(kuberay_amd/utils/fake_dashboard.py) through the same interface — the
ClientProvider seam of the reference (suite_test.go:57-69).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

import httpx


class DashboardClientError(Exception):
    pass


class RayDashboardClientInterface:
    """dashboard_httpclient.go:29-41."""

    # serve
    def get_serve_applications(self) -> Dict[str, Any]:
        raise NotImplementedError

    def update_serve_applications(self, config: Dict[str, Any]) -> None:
        raise NotImplementedError

    # jobs
    def get_job_info(self, job_id: str) -> Optional[Dict[str, Any]]:
        raise NotImplementedError

    def list_jobs(self) -> List[Dict[str, Any]]:
        raise NotImplementedError

    def submit_job(self, submission: Dict[str, Any]) -> str:
        raise NotImplementedError

    def stop_job(self, job_id: str) -> None:
        raise NotImplementedError

    def delete_job(self, job_id: str) -> None:
        raise NotImplementedError

    def get_job_log(self, job_id: str) -> str:
        raise NotImplementedError


class RayDashboardClient(RayDashboardClientInterface):
    def __init__(self, dashboard_url: str, auth_token: Optional[str] = None,
                 timeout: float = 5.0):
        if not dashboard_url.startswith("http"):
            dashboard_url = "http://" + dashboard_url
        self.base = dashboard_url.rstrip("/")
        headers = {}
        if auth_token:
            headers["Authorization"] = f"Bearer {auth_token}"
        self._http = httpx.Client(base_url=self.base, timeout=timeout, headers=headers)

    def _check(self, resp: httpx.Response) -> httpx.Response:
        if resp.status_code >= 400:
            raise DashboardClientError(
                f"{resp.request.method} {resp.request.url} -> {resp.status_code}: {resp.text[:500]}")
        return resp

    # -- serve (GET/PUT /api/serve/applications/) ----------------------
    def get_serve_applications(self) -> Dict[str, Any]:
        return self._check(self._http.get("/api/serve/applications/")).json()

    def update_serve_applications(self, config: Dict[str, Any]) -> None:
        self._check(self._http.put("/api/serve/applications/", json=config))

    # -- jobs (/api/jobs/) ---------------------------------------------
    def get_job_info(self, job_id: str) -> Optional[Dict[str, Any]]:
        resp = self._http.get(f"/api/jobs/{job_id}")
        if resp.status_code == 404:
            return None
        return self._check(resp).json()

    def list_jobs(self) -> List[Dict[str, Any]]:
        return self._check(self._http.get("/api/jobs/")).json()

    def submit_job(self, submission: Dict[str, Any]) -> str:
        resp = self._check(self._http.post("/api/jobs/", json=submission))
        return resp.json().get("submission_id", "")

    def stop_job(self, job_id: str) -> None:
        self._check(self._http.post(f"/api/jobs/{job_id}/stop"))

    def delete_job(self, job_id: str) -> None:
        self._check(self._http.delete(f"/api/jobs/{job_id}"))

    def get_job_log(self, job_id: str) -> str:
        return self._check(self._http.get(f"/api/jobs/{job_id}/logs")).json().get("logs", "")


class RayHttpProxyClient:
    """Serve proxy healthz check (reference: utils/httpproxy_httpclient.go) —
    used by the RayService controller to flip the head pod's serve label."""

    def __init__(self, timeout: float = 2.0):
        self._http = httpx.Client(timeout=timeout)

    def check_proxy_healthy(self, pod_ip: str, port: int = 8000) -> bool:
        try:
            resp = self._http.get(f"http://{pod_ip}:{port}/-/healthz")
            return resp.status_code == 200
        except httpx.HTTPError:
            return False
