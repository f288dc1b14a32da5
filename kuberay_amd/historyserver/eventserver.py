"""Event replay → job/task/actor/node state
(reference: historyserver/pkg/eventserver/eventserver.go:67-1149 — the
lifecycle state machines that turn stored Ray events back into
dashboard-shaped state, including the task-timeline reconstruction of
historyserver/pkg/historyserver/timeline.go).
"""
from __future__ import annotations

import json
import logging
from typing import Any, Dict, List

from .storage import StorageReader, decompress

logger = logging.getLogger("kuberay.historyserver")

# Ray event types (reference DEFAULT_RAY_EXPOSABLE_EVENT_TYPES)
TASK_DEFINITION = "TASK_DEFINITION_EVENT"
TASK_LIFECYCLE = "TASK_LIFECYCLE_EVENT"
TASK_PROFILE = "TASK_PROFILE_EVENT"
ACTOR_DEFINITION = "ACTOR_DEFINITION_EVENT"
ACTOR_LIFECYCLE = "ACTOR_LIFECYCLE_EVENT"
NODE_DEFINITION = "NODE_DEFINITION_EVENT"
NODE_LIFECYCLE = "NODE_LIFECYCLE_EVENT"
JOB_DEFINITION = "DRIVER_JOB_DEFINITION_EVENT"
JOB_LIFECYCLE = "DRIVER_JOB_LIFECYCLE_EVENT"


class SessionState:
    """Accumulated dashboard-shaped state for one cluster session."""

    def __init__(self) -> None:
        self.jobs: Dict[str, Dict[str, Any]] = {}
        self.tasks: Dict[str, Dict[str, Any]] = {}
        self.actors: Dict[str, Dict[str, Any]] = {}
        self.nodes: Dict[str, Dict[str, Any]] = {}
        self.profile_events: List[Dict[str, Any]] = []

    # -- handlers (eventserver.go handleXxx analogs) --------------------
    def apply(self, event: Dict[str, Any]) -> None:
        etype = event.get("event_type") or event.get("eventType") or ""
        data = event.get("data") or event
        handler = {
            JOB_DEFINITION: self._job_def, JOB_LIFECYCLE: self._job_lc,
            TASK_DEFINITION: self._task_def, TASK_LIFECYCLE: self._task_lc,
            ACTOR_DEFINITION: self._actor_def, ACTOR_LIFECYCLE: self._actor_lc,
            NODE_DEFINITION: self._node_def, NODE_LIFECYCLE: self._node_lc,
            TASK_PROFILE: self._task_profile,
        }.get(etype)
        if handler:
            handler(data)

    def _job_def(self, d):
        job_id = d.get("job_id") or d.get("jobId", "")
        self.jobs.setdefault(job_id, {"job_id": job_id}).update({
            "entrypoint": d.get("entrypoint"),
            "submission_id": d.get("submission_id"),
        })

    def _job_lc(self, d):
        job_id = d.get("job_id") or d.get("jobId", "")
        job = self.jobs.setdefault(job_id, {"job_id": job_id})
        state = d.get("state") or d.get("status")
        if state:
            job["status"] = state
        for k_src, k_dst in (("start_time", "start_time"), ("end_time", "end_time")):
            if d.get(k_src):
                job[k_dst] = d[k_src]

    def _task_def(self, d):
        task_id = d.get("task_id") or d.get("taskId", "")
        self.tasks.setdefault(task_id, {"task_id": task_id}).update({
            "name": d.get("name") or d.get("func_or_class_name"),
            "job_id": d.get("job_id"),
            "actor_id": d.get("actor_id"),
        })

    def _task_lc(self, d):
        task_id = d.get("task_id") or d.get("taskId", "")
        task = self.tasks.setdefault(task_id, {"task_id": task_id})
        state = d.get("state") or d.get("status")
        if state:
            task["state"] = state
        ts = d.get("timestamp") or d.get("ts")
        if ts is not None:
            task.setdefault("state_ts", {})[state or "?"] = ts

    def _actor_def(self, d):
        actor_id = d.get("actor_id") or d.get("actorId", "")
        self.actors.setdefault(actor_id, {"actor_id": actor_id}).update({
            "class_name": d.get("class_name") or d.get("name"),
            "job_id": d.get("job_id"),
        })

    def _actor_lc(self, d):
        actor_id = d.get("actor_id") or d.get("actorId", "")
        actor = self.actors.setdefault(actor_id, {"actor_id": actor_id})
        state = d.get("state") or d.get("status")
        if state:
            actor["state"] = state

    def _node_def(self, d):
        node_id = d.get("node_id") or d.get("nodeId", "")
        self.nodes.setdefault(node_id, {"node_id": node_id}).update({
            "node_ip": d.get("node_ip") or d.get("nodeIp"),
            "resources": d.get("resources"),
        })

    def _node_lc(self, d):
        node_id = d.get("node_id") or d.get("nodeId", "")
        node = self.nodes.setdefault(node_id, {"node_id": node_id})
        state = d.get("state") or d.get("status")
        if state:
            node["state"] = state

    def _task_profile(self, d):
        self.profile_events.append(d)

    # -- timeline (timeline.go analog; trace-event-format output) -------
    def timeline(self) -> List[Dict[str, Any]]:
        """Chrome trace-viewer events from profile + task lifecycle data."""
        out = []
        for ev in self.profile_events:
            start = ev.get("start_time") or ev.get("startTime")
            end = ev.get("end_time") or ev.get("endTime")
            if start is None:
                continue
            entry = {
                "name": ev.get("event_name") or ev.get("name", "task"),
                "cat": ev.get("component_type", "task"),
                "pid": ev.get("node_ip_address", "node"),
                "tid": ev.get("component_id", "worker"),
                "ts": float(start) * 1e6,
                "ph": "X" if end is not None else "B",
            }
            if end is not None:
                entry["dur"] = max(0.0, (float(end) - float(start))) * 1e6
            out.append(entry)
        return sorted(out, key=lambda e: e["ts"])


def load_session(storage: StorageReader, prefix: str) -> SessionState:
    """Replay every stored event batch under {prefix}/events/."""
    state = SessionState()
    for path in storage.list(f"{prefix}/events"):
        try:
            raw = storage.read(path)
            if path.endswith(".gz"):
                raw = decompress(raw)
            text = raw.decode()
        except Exception:
            # one corrupt batch must not kill post-mortem browsing of the
            # rest of the session
            logger.warning("skipping unreadable event batch %s", path)
            continue
        for line in text.splitlines():
            line = line.strip()
            if not line:
                continue
            try:
                state.apply(json.loads(line))
            except ValueError:
                continue
    return state
