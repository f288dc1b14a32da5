"""Fake Ray dashboard client (reference: utils/fake_serve_httpclient.go:14
FakeRayDashboardClient + fake_httpproxy_httpclient.go).

Deterministic in-memory behavior with mockable hooks:

* jobs auto-advance PENDING → RUNNING → SUCCEEDED after a configurable
  number of polls (or use ``set_job_status`` / ``get_job_info_mock``),
* serve applications become RUNNING ``serve_deploy_delay_polls`` polls after
  the last config PUT.
"""
from __future__ import annotations

import threading
from typing import Any, Callable, Dict, List, Optional

import yaml

from ..models.rayservice import ApplicationStatus
from .dashboard_client import RayDashboardClientInterface


class FakeRayDashboardClient(RayDashboardClientInterface):
    def __init__(self, job_polls_to_running: int = 1, job_polls_to_succeeded: int = 3,
                 serve_deploy_delay_polls: int = 1):
        self._lock = threading.Lock()
        self.jobs: Dict[str, Dict[str, Any]] = {}
        self._job_polls: Dict[str, int] = {}
        self.job_polls_to_running = job_polls_to_running
        self.job_polls_to_succeeded = job_polls_to_succeeded
        self.get_job_info_mock: Optional[Callable[[str], Optional[Dict[str, Any]]]] = None

        self.serve_config: Optional[Dict[str, Any]] = None
        self._serve_polls_since_deploy = 0
        self.serve_deploy_delay_polls = serve_deploy_delay_polls
        self.serve_statuses_mock: Optional[Dict[str, Any]] = None
        self.update_serve_calls: List[Dict[str, Any]] = []
        self.stopped_jobs: List[str] = []
        self.deleted_jobs: List[str] = []

    # -- manual-control hooks ------------------------------------------
    def set_job_status(self, job_id: str, status: str) -> None:
        with self._lock:
            self.jobs.setdefault(job_id, {"submission_id": job_id})["status"] = status

    # -- jobs ----------------------------------------------------------
    def get_job_info(self, job_id: str) -> Optional[Dict[str, Any]]:
        if self.get_job_info_mock is not None:
            return self.get_job_info_mock(job_id)
        with self._lock:
            job = self.jobs.get(job_id)
            if job is None:
                # auto-register: the submitter (sim kubelet job) is assumed to
                # have submitted by the time the controller polls.
                job = {"submission_id": job_id, "status": "PENDING",
                       "start_time": 1, "end_time": 0}
                self.jobs[job_id] = job
            if job.get("status") not in ("STOPPED", "SUCCEEDED", "FAILED", "_pinned"):
                polls = self._job_polls.get(job_id, 0) + 1
                self._job_polls[job_id] = polls
                if polls >= self.job_polls_to_succeeded:
                    job["status"] = "SUCCEEDED"
                    job["end_time"] = 2
                elif polls >= self.job_polls_to_running:
                    job["status"] = "RUNNING"
            return dict(job)

    def list_jobs(self) -> List[Dict[str, Any]]:
        with self._lock:
            return [dict(j) for j in self.jobs.values()]

    def submit_job(self, submission: Dict[str, Any]) -> str:
        job_id = submission.get("submission_id", f"job-{len(self.jobs)+1}")
        with self._lock:
            self.jobs[job_id] = {"submission_id": job_id, "status": "PENDING"}
        return job_id

    def stop_job(self, job_id: str) -> None:
        with self._lock:
            self.stopped_jobs.append(job_id)
            if job_id in self.jobs:
                self.jobs[job_id]["status"] = "STOPPED"

    def delete_job(self, job_id: str) -> None:
        with self._lock:
            self.deleted_jobs.append(job_id)
            self.jobs.pop(job_id, None)

    def get_job_log(self, job_id: str) -> str:
        return ""

    # -- serve ---------------------------------------------------------
    def update_serve_applications(self, config: Dict[str, Any]) -> None:
        with self._lock:
            self.serve_config = config
            self.update_serve_calls.append(config)
            self._serve_polls_since_deploy = 0

    def get_serve_applications(self) -> Dict[str, Any]:
        if self.serve_statuses_mock is not None:
            return self.serve_statuses_mock
        with self._lock:
            if self.serve_config is None:
                return {"applications": {}}
            self._serve_polls_since_deploy += 1
            ready = self._serve_polls_since_deploy > self.serve_deploy_delay_polls
            status = (ApplicationStatus.RUNNING if ready
                      else ApplicationStatus.DEPLOYING)
            apps = {}
            for app in self.serve_config.get("applications", []):
                name = app.get("name", "default")
                deployments = {
                    d.get("name", f"d{i}"): {"status": "HEALTHY" if ready else "UPDATING",
                                             "message": ""}
                    for i, d in enumerate(app.get("deployments", []) or
                                          [{"name": "default"}])
                }
                apps[name] = {
                    "status": status,
                    "message": "",
                    "deployments": deployments,
                    "route_prefix": app.get("route_prefix", "/"),
                }
            return {"applications": apps}


class FakeRayHttpProxyClient:
    """fake_httpproxy_httpclient.go analog — always healthy unless told not."""

    def __init__(self, healthy: bool = True):
        self.healthy = healthy

    def check_proxy_healthy(self, pod_ip: str, port: int = 8000) -> bool:
        return self.healthy


def parse_serve_config_v2(serve_config_v2: str) -> Dict[str, Any]:
    data = yaml.safe_load(serve_config_v2) if serve_config_v2 else {}
    return data or {}
