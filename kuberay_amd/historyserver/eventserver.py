"""Event replay → job/task/actor/node state machines.

Reference: historyserver/pkg/eventserver/eventserver.go:67-1149 (the
lifecycle state machines that turn stored Ray events back into
dashboard-shaped state), state_transition.go (dedup/sort merge),
types/{task,actor,job,node}.go (derived-field rules) and
historyserver/pkg/historyserver/timeline.go:13-252 (Chrome-trace
reconstruction).

Events arrive in Ray's export-event envelope::

    {"eventType": "TASK_LIFECYCLE_EVENT", "taskLifecycleEvent": {...}}

with IDs base64-encoded; replay normalizes them to hex exactly like the
reference's ConvertBase64ToHex. The older flat format used by round-1
fixtures ({"event_type": ..., fields inline}) is still accepted.
"""
from __future__ import annotations

import base64
import binascii
import json
import logging
from typing import Any, Dict, List, Optional, Tuple

from .storage import StorageReader, decompress

logger = logging.getLogger("kuberay.historyserver")

# Ray export event types (types/event.go)
TASK_DEFINITION = "TASK_DEFINITION_EVENT"
TASK_LIFECYCLE = "TASK_LIFECYCLE_EVENT"
ACTOR_TASK_DEFINITION = "ACTOR_TASK_DEFINITION_EVENT"
TASK_PROFILE = "TASK_PROFILE_EVENT"
ACTOR_DEFINITION = "ACTOR_DEFINITION_EVENT"
ACTOR_LIFECYCLE = "ACTOR_LIFECYCLE_EVENT"
NODE_DEFINITION = "NODE_DEFINITION_EVENT"
NODE_LIFECYCLE = "NODE_LIFECYCLE_EVENT"
JOB_DEFINITION = "DRIVER_JOB_DEFINITION_EVENT"
JOB_LIFECYCLE = "DRIVER_JOB_LIFECYCLE_EVENT"

# task states with a derived-time meaning (types/task.go:70-86 + common.py)
_TASK_CREATED_STATE = "PENDING_ARGS_AVAIL"
_TASK_RUNNING_STATE = "RUNNING"
_TASK_END_STATES = {"FINISHED", "FAILED"}

_HEX_DIGITS = set("0123456789abcdef")


def normalize_id(value: Optional[str]) -> str:
    """ConvertBase64ToHex analog (utils.ConvertBase64ToHex).

    Ray export events carry binary IDs base64-encoded; the dashboard (and
    this server) keys everything by the hex form. Values that already look
    like hex are kept (idempotent re-replay of normalized snapshots).
    """
    if not value:
        return ""
    lowered = value.lower()
    if set(lowered) <= _HEX_DIGITS and len(value) % 2 == 0:
        return lowered
    try:
        raw = base64.b64decode(value, validate=True)
    except (binascii.Error, ValueError):
        return value
    return raw.hex()


def extract_actor_id_from_task_id(task_id_hex: str) -> str:
    """timeline.go:231-252 — TaskID = 8B unique + 16B ActorID; the actor
    portion all-Fs means a normal/driver task with no actor."""
    if len(task_id_hex) != 48:
        return ""
    actor_portion = task_id_hex[16:40]
    job_portion = task_id_hex[40:48]
    if actor_portion.lower() == "f" * 24:
        return ""
    return actor_portion + job_portion


def _ts_key(ts: Any) -> float:
    """Sortable key for RFC3339 / epoch-seconds / epoch-nanos timestamps."""
    if ts is None:
        return 0.0
    if isinstance(ts, (int, float)):
        return float(ts)
    try:
        return float(ts)
    except (TypeError, ValueError):
        pass
    import datetime as dt
    try:
        return dt.datetime.fromisoformat(
            str(ts).replace("Z", "+00:00")).timestamp()
    except ValueError:
        return 0.0


def merge_state_transitions(existing: List[Dict[str, Any]],
                            new: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
    """state_transition.go MergeStateTransitions: dedup on
    (state, timestamp), chronological sort."""
    seen = {(t.get("state"), str(t.get("timestamp"))) for t in existing}
    merged = list(existing)
    for t in new:
        key = (t.get("state"), str(t.get("timestamp")))
        if key not in seen:
            merged.append(t)
            seen.add(key)
    merged.sort(key=lambda t: _ts_key(t.get("timestamp")))
    return merged


def _merge_task_log_stream(cur: Dict[str, Any], upd: Dict[str, Any],
                           file_key: str, start_key: str, end_key: str) -> None:
    """mergeTaskLogStream (eventserver.go:769-786): Ray sends log-start and
    log-end as separate partial updates."""
    if upd.get(file_key):
        cur[file_key] = upd[file_key]
        cur[start_key] = upd.get(start_key, 0)
    elif upd.get(start_key):
        cur[start_key] = upd[start_key]
    if upd.get(end_key) or not cur.get(end_key):
        cur[end_key] = upd.get(end_key, 0)


def merge_task_log_info(current: Optional[Dict[str, Any]],
                        update: Optional[Dict[str, Any]]) -> Optional[Dict[str, Any]]:
    """mergeTaskLogInfo (eventserver.go:741-767)."""
    if update is None:
        return current
    if current is None:
        return dict(update)
    _merge_task_log_stream(current, update, "stdoutFile", "stdoutStart",
                           "stdoutEnd")
    _merge_task_log_stream(current, update, "stderrFile", "stderrStart",
                           "stderrEnd")
    return current


class SessionState:
    """Accumulated dashboard-shaped state for one cluster session.

    ``tasks`` is keyed by ``"{task_id}:{attempt}"`` — the reference keeps a
    per-task attempt map (types/task.go TaskMap.CreateOrMergeAttempt) and
    the dashboard lists every attempt as its own row.
    """

    def __init__(self) -> None:
        self.jobs: Dict[str, Dict[str, Any]] = {}
        self.tasks: Dict[str, Dict[str, Any]] = {}
        self.actors: Dict[str, Dict[str, Any]] = {}
        self.nodes: Dict[str, Dict[str, Any]] = {}
        # /events API: log events grouped by job (types/log_event.go)
        self.log_events: Dict[str, List[Dict[str, Any]]] = {}

    # -- dispatch (eventserver.go storeEvent :125-521) -------------------
    def apply(self, event: Dict[str, Any]) -> None:
        etype = event.get("eventType") or event.get("event_type") or ""
        if etype == TASK_DEFINITION:
            self._task_definition(
                event.get("taskDefinitionEvent") or event.get("data") or event,
                is_actor_task=False)
        elif etype == ACTOR_TASK_DEFINITION:
            self._task_definition(
                event.get("actorTaskDefinitionEvent") or event.get("data")
                or event, is_actor_task=True)
        elif etype == TASK_LIFECYCLE:
            self._task_lifecycle(
                event.get("taskLifecycleEvent") or event.get("data") or event)
        elif etype == TASK_PROFILE:
            self._task_profile(
                event.get("taskProfileEvents") or event.get("data") or event)
        elif etype == ACTOR_DEFINITION:
            self._actor_definition(
                event.get("actorDefinitionEvent") or event.get("data") or event)
        elif etype == ACTOR_LIFECYCLE:
            self._actor_lifecycle(
                event.get("actorLifecycleEvent") or event.get("data") or event)
        elif etype == NODE_DEFINITION:
            self._node_definition(
                event.get("nodeDefinitionEvent") or event.get("data") or event)
        elif etype == NODE_LIFECYCLE:
            self._node_lifecycle(
                event.get("nodeLifecycleEvent") or event.get("data") or event)
        elif etype == JOB_DEFINITION:
            self._job_definition(
                event.get("driverJobDefinitionEvent") or event.get("data")
                or event)
        elif etype == JOB_LIFECYCLE:
            self._job_lifecycle(
                event.get("driverJobLifecycleEvent") or event.get("data")
                or event)
        else:
            logger.debug("event type not supported, skipping: %s", etype)

    # -- tasks (eventserver.go:658-884) ----------------------------------
    @staticmethod
    def _field(d: Dict[str, Any], *names, default=None):
        for n in names:
            if d.get(n) is not None:
                return d[n]
        return default

    def _task_key(self, task_id: str, attempt: int) -> str:
        return f"{task_id}:{attempt}"

    def _get_or_create_attempt(self, task_id: str, attempt: int) -> Dict[str, Any]:
        key = self._task_key(task_id, attempt)
        return self.tasks.setdefault(key, {
            "taskId": task_id, "taskAttempt": attempt,
            "stateTransitions": []})

    def _task_definition(self, d: Dict[str, Any], is_actor_task: bool) -> None:
        task_id = normalize_id(self._field(d, "taskId", "task_id", default=""))
        if not task_id:
            return
        attempt = int(self._field(d, "taskAttempt", "task_attempt", default=0))
        task = self._get_or_create_attempt(task_id, attempt)
        # definition fields overwrite; lifecycle/profile-derived fields are
        # preserved (handleTaskDefinitionEvent :692-738)
        func = self._field(d, "taskFunc", "funcOrClassName",
                           "func_or_class_name")
        if isinstance(func, dict):
            # FunctionDescriptor union (types/task.go:54-61)
            pfd = func.get("pythonFunctionDescriptor") or {}
            name_parts = [pfd.get("className"), pfd.get("functionName")]
            func = ".".join(p for p in name_parts if p) or None
        task.update({
            "taskType": ("ACTOR_TASK" if is_actor_task
                         else self._field(d, "taskType", "task_type",
                                          default="NORMAL_TASK")),
            "language": self._field(d, "language", default="PYTHON"),
            "requiredResources": self._field(d, "requiredResources",
                                             "required_resources"),
            "runtimeEnvInfo": self._field(d, "runtimeEnvInfo"),
            "placementGroupId": normalize_id(self._field(
                d, "placementGroupId", "placement_group_id", default="")),
            "parentTaskId": normalize_id(self._field(
                d, "parentTaskId", "parent_task_id", default="")),
        })
        if func and not (task.get("profileData") and task.get("funcOrClassName")):
            task["funcOrClassName"] = func
        job_id = normalize_id(self._field(d, "jobId", "job_id", default=""))
        if job_id:
            task["jobId"] = job_id
        actor_id = normalize_id(self._field(d, "actorId", "actor_id",
                                            default=""))
        if actor_id:
            task["actorId"] = actor_id

    def _task_lifecycle(self, d: Dict[str, Any]) -> None:
        task_id = normalize_id(self._field(d, "taskId", "task_id", default=""))
        if not task_id:
            return
        attempt = int(self._field(d, "taskAttempt", "task_attempt", default=0))
        transitions = [
            {"state": t.get("state"), "timestamp": t.get("timestamp")}
            for t in (self._field(d, "stateTransitions", "state_transitions")
                      or [])]
        # flat round-1 format: single state + ts inline
        if not transitions and (d.get("state") or d.get("status")):
            transitions = [{"state": d.get("state") or d.get("status"),
                            "timestamp": d.get("timestamp") or d.get("ts")}]
        log_info = self._field(d, "taskLogInfo", "task_log_info")
        if not transitions and log_info is None:
            return  # :810-812 — must carry a transition or log info
        task = self._get_or_create_attempt(task_id, attempt)
        task["stateTransitions"] = merge_state_transitions(
            task["stateTransitions"], transitions)
        job_id = normalize_id(self._field(d, "jobId", "job_id", default=""))
        if job_id:
            task["jobId"] = job_id
        task["taskLogInfo"] = merge_task_log_info(task.get("taskLogInfo"),
                                                  log_info)
        if not transitions:
            return
        if self._field(d, "rayErrorInfo", "ray_error_info") is not None:
            task["rayErrorInfo"] = self._field(d, "rayErrorInfo",
                                               "ray_error_info")
        for src, dst in (("nodeId", "nodeId"), ("workerId", "workerId")):
            val = normalize_id(self._field(d, src, default=""))
            if val:
                task[dst] = val
        pid = self._field(d, "workerPid", "worker_pid")
        if pid:
            task["workerPid"] = int(pid)
        if self._field(d, "isDebuggerPaused") is not None:
            task["isDebuggerPaused"] = bool(d.get("isDebuggerPaused"))
        if self._field(d, "actorReprName"):
            task["actorReprName"] = d.get("actorReprName")
        task["state"] = (task["stateTransitions"][-1]["state"]
                         if task["stateTransitions"] else "NIL")
        # derived times (:850-868; ray state common.py:1660-1685)
        for tr in task["stateTransitions"]:
            state, ts = tr.get("state"), tr.get("timestamp")
            if state == _TASK_CREATED_STATE and not task.get("creationTime"):
                task["creationTime"] = ts
            elif state == _TASK_RUNNING_STATE and not task.get("startTime"):
                task["startTime"] = ts
            elif state in _TASK_END_STATES:
                task["endTime"] = ts

    def _task_profile(self, d: Dict[str, Any]) -> None:
        """handleTaskProfileEvent (:967-1080)."""
        task_id = normalize_id(self._field(d, "taskId", "task_id", default=""))
        profile = self._field(d, "profileEvents", "profile_events", default={})
        events_in = (profile.get("events") if isinstance(profile, dict)
                     else None) or d.get("events") or []
        if not task_id or not events_in:
            return
        attempt = int(self._field(d, "attemptNumber", "attempt_number",
                                  "taskAttempt", default=0))
        raw_events = []
        for e in events_in:
            try:
                start = int(e.get("startTime") or e.get("start_time"))
                end = int(e.get("endTime") or e.get("end_time"))
            except (TypeError, ValueError):
                continue
            raw_events.append({
                "eventName": e.get("eventName") or e.get("event_name") or "",
                "startTime": start, "endTime": end,
                "extraData": e.get("extraData") or e.get("extra_data") or ""})
        task = self._get_or_create_attempt(task_id, attempt)
        job_id = normalize_id(self._field(d, "jobId", "job_id", default=""))
        if job_id and not task.get("jobId"):
            task["jobId"] = job_id
        pd = task.setdefault("profileData", {
            "componentId": normalize_id(
                profile.get("componentId", "") if isinstance(profile, dict)
                else ""),
            "componentType": (profile.get("componentType", "")
                              if isinstance(profile, dict) else ""),
            "nodeIpAddress": (profile.get("nodeIpAddress", "")
                              if isinstance(profile, dict) else ""),
            "events": []})
        seen = {(e["eventName"], e["startTime"], e["endTime"])
                for e in pd["events"]}
        for e in raw_events:
            key = (e["eventName"], e["startTime"], e["endTime"])
            if key not in seen:
                pd["events"].append(e)
                seen.add(key)
        # func_or_class_name from the overall task::X event (:1060-1077)
        for e in raw_events:
            if e["eventName"].startswith("task::") and e["extraData"]:
                try:
                    extra = json.loads(e["extraData"])
                except ValueError:
                    continue
                if extra.get("name"):
                    task["funcOrClassName"] = e["eventName"][len("task::"):]

    # -- actors (eventserver.go:146-344) ----------------------------------
    def _actor_definition(self, d: Dict[str, Any]) -> None:
        actor_id = normalize_id(self._field(d, "actorId", "actor_id",
                                            default=""))
        if not actor_id:
            return
        actor = self.actors.setdefault(actor_id, {"actorId": actor_id,
                                                  "events": []})
        # definition overwrites; lifecycle-derived fields preserved
        actor.update({
            "jobId": normalize_id(self._field(d, "jobId", "job_id",
                                              default="")) or actor.get("jobId"),
            "className": self._field(d, "className", "class_name", "name"),
            "name": self._field(d, "name", default=actor.get("name")),
            "rayNamespace": self._field(d, "rayNamespace"),
            "isDetached": self._field(d, "isDetached"),
            "requiredResources": self._field(d, "requiredResources",
                                             "required_resources"),
            "placementGroupId": normalize_id(self._field(
                d, "placementGroupId", "placement_group_id", default="")),
        })

    def _actor_lifecycle(self, d: Dict[str, Any]) -> None:
        actor_id = normalize_id(self._field(d, "actorId", "actor_id",
                                            default=""))
        transitions = self._field(d, "stateTransitions", "state_transitions") \
            or []
        if not transitions and (d.get("state") or d.get("status")):
            transitions = [{"state": d.get("state") or d.get("status"),
                            "timestamp": d.get("timestamp")}]
        if not actor_id or not transitions:
            return
        events = []
        for tr in transitions:
            death = tr.get("deathCause")
            events.append({
                "state": tr.get("state"),
                "timestamp": tr.get("timestamp"),
                "nodeId": normalize_id(tr.get("nodeId", "")),
                "workerId": normalize_id(tr.get("workerId", "")),
                "reprName": tr.get("reprName", ""),
                "deathCause": (json.dumps(death) if isinstance(death, dict)
                               else (death or "")),
            })
        actor = self.actors.setdefault(actor_id, {"actorId": actor_id,
                                                  "events": []})
        actor["events"] = merge_state_transitions(actor.get("events", []),
                                                  events)
        if not actor["events"]:
            return
        last = actor["events"][-1]
        actor["state"] = last["state"]
        # address from the most recent ALIVE transition (:283-290)
        for e in reversed(actor["events"]):
            if e["state"] == "ALIVE" and e.get("nodeId"):
                actor.setdefault("address", {})
                actor["address"]["nodeId"] = e["nodeId"]
                actor["address"]["workerId"] = e.get("workerId", "")
                break
        if last.get("reprName"):
            actor["reprName"] = last["reprName"]
        if not actor.get("startTime"):
            for e in actor["events"]:
                if e["state"] == "ALIVE":
                    actor["startTime"] = e["timestamp"]
                    break
        if last["state"] == "DEAD":
            actor["endTime"] = last["timestamp"]
            if last.get("deathCause"):
                try:
                    ctx = json.loads(last["deathCause"]).get(
                        "actorDiedErrorContext") or {}
                except ValueError:
                    ctx = {}
                if ctx.get("pid") is not None:
                    actor["pid"] = int(ctx["pid"])
                if ctx.get("nodeIpAddress"):
                    actor.setdefault("address", {})["ipAddress"] = \
                        ctx["nodeIpAddress"]
                if ctx.get("errorMessage"):
                    actor["exitDetails"] = ctx["errorMessage"]
        actor["numRestarts"] = sum(1 for e in actor["events"]
                                   if e["state"] == "RESTARTING")

    # -- jobs (eventserver.go:346-499) -------------------------------------
    def _job_definition(self, d: Dict[str, Any]) -> None:
        job_id = normalize_id(self._field(d, "jobId", "job_id", default=""))
        if not job_id:
            return
        job = self.jobs.setdefault(job_id, {"jobId": job_id,
                                            "stateTransitions": []})
        job.update({
            "entrypoint": self._field(d, "entrypoint",
                                      default=job.get("entrypoint")),
            "submissionId": self._field(d, "submissionId", "submission_id",
                                        default=job.get("submissionId")),
            "driverNodeId": normalize_id(self._field(
                d, "driverNodeId", "driver_node_id", default=""))
            or job.get("driverNodeId"),
            "driverPid": self._field(d, "driverPid",
                                     default=job.get("driverPid")),
            "config": self._field(d, "config", default=job.get("config")),
        })

    def _job_lifecycle(self, d: Dict[str, Any]) -> None:
        job_id = normalize_id(self._field(d, "jobId", "job_id", default=""))
        transitions = self._field(d, "stateTransitions", "state_transitions") \
            or []
        if not transitions and (d.get("state") or d.get("status")):
            transitions = [{"state": d.get("state") or d.get("status"),
                            "timestamp": d.get("timestamp")}]
        if not job_id or not transitions:
            return
        job = self.jobs.setdefault(job_id, {"jobId": job_id,
                                            "stateTransitions": []})
        job["stateTransitions"] = merge_state_transitions(
            job["stateTransitions"],
            [{"state": t.get("state"), "timestamp": t.get("timestamp")}
             for t in transitions])
        if not job["stateTransitions"]:
            return
        last = job["stateTransitions"][-1]
        job["state"] = last["state"]
        if not job.get("startTime"):
            for t in job["stateTransitions"]:
                if t["state"] == "CREATED":
                    job["startTime"] = t["timestamp"]
                    break
        if last["state"] == "FINISHED":
            job["endTime"] = last["timestamp"]

    # -- nodes (eventserver.go:886-965) -------------------------------------
    def _node_definition(self, d: Dict[str, Any]) -> None:
        node_id = normalize_id(self._field(d, "nodeId", "node_id", default=""))
        if not node_id:
            return
        node = self.nodes.setdefault(node_id, {"nodeId": node_id,
                                               "stateTransitions": []})
        node.update({
            "nodeIpAddress": self._field(d, "nodeIpAddress", "node_ip",
                                         "nodeIp",
                                         default=node.get("nodeIpAddress")),
            "startTimestamp": self._field(d, "startTimestamp",
                                          default=node.get("startTimestamp")),
            "labels": self._field(d, "labels", default=node.get("labels")),
            "resources": self._field(d, "resources",
                                     default=node.get("resources")),
        })

    def _node_lifecycle(self, d: Dict[str, Any]) -> None:
        node_id = normalize_id(self._field(d, "nodeId", "node_id", default=""))
        transitions = self._field(d, "stateTransitions", "state_transitions") \
            or []
        if not transitions and (d.get("state") or d.get("status")):
            transitions = [{"state": d.get("state") or d.get("status"),
                            "timestamp": d.get("timestamp")}]
        if not node_id or not transitions:
            return
        node = self.nodes.setdefault(node_id, {"nodeId": node_id,
                                               "stateTransitions": []})
        node["stateTransitions"] = merge_state_transitions(
            node["stateTransitions"],
            [{"state": t.get("state"), "timestamp": t.get("timestamp"),
              **({"resources": t["resources"]} if t.get("resources") else {}),
              **({"deathInfo": t["deathInfo"]} if t.get("deathInfo") else {})}
             for t in transitions])
        if node["stateTransitions"]:
            last = node["stateTransitions"][-1]
            node["state"] = last["state"]
            # ALIVE carries resources; DEAD carries deathInfo (node.go:31-46)
            for t in node["stateTransitions"]:
                if t.get("resources"):
                    node["resources"] = t["resources"]
            if last["state"] == "DEAD":
                node["endTime"] = last["timestamp"]
                if last.get("deathInfo"):
                    node["deathInfo"] = last["deathInfo"]

    # -- log events (/events API; log_event_reader.go) ----------------------
    def apply_log_event(self, event: Dict[str, Any]) -> None:
        """One line of logs/{node}/events/event_*.log (Ray dashboard
        event_utils.py format): label/message/timestamp/severity +
        custom_fields carrying job_id."""
        custom = event.get("custom_fields") or {}
        job_id = (event.get("job_id") or custom.get("job_id") or "global")
        self.log_events.setdefault(job_id, []).append({
            "eventId": event.get("event_id") or event.get("eventId"),
            "sourceType": event.get("source_type") or event.get("sourceType"),
            "hostName": event.get("host_name") or event.get("hostName"),
            "pid": event.get("pid"),
            "label": event.get("label", ""),
            "message": event.get("message", ""),
            "timestamp": event.get("timestamp"),
            "severity": event.get("severity", "INFO"),
            "customFields": custom,
        })

    # -- timeline (timeline.go:13-252) --------------------------------------
    def timeline(self, job_id: Optional[str] = None) -> List[Dict[str, Any]]:
        """Chrome-trace events matching Ray Dashboard's
        /api/v0/tasks/timeline: process/thread metadata rows + one "X" slice
        per profile event, colored like Ray's profiling.py."""
        tasks = [t for t in self.tasks.values()
                 if (job_id is None or t.get("jobId") == job_id)]
        filtered = []
        for t in tasks:
            pd = t.get("profileData")
            if not pd or not pd.get("events"):
                continue
            if pd.get("componentType") not in ("worker", "driver"):
                continue
            if not pd.get("nodeIpAddress"):
                continue
            filtered.append(t)
        if not filtered:
            return []

        node_ip_to_pid: Dict[str, int] = {}
        tid_map: Dict[Tuple[str, str], int] = {}
        for t in filtered:
            pd = t["profileData"]
            ip = pd["nodeIpAddress"]
            comp = f"{pd.get('componentType', '')}:{pd.get('componentId', '')}"
            if ip not in node_ip_to_pid:
                node_ip_to_pid[ip] = len(node_ip_to_pid)
            tid_map.setdefault((ip, comp), len(tid_map))

        events: List[Dict[str, Any]] = []
        for ip, pid in node_ip_to_pid.items():
            events.append({"name": "process_name", "pid": pid, "tid": None,
                           "ph": "M", "args": {"name": f"Node {ip}"}})
        for (ip, comp), tid in tid_map.items():
            events.append({"name": "thread_name",
                           "pid": node_ip_to_pid[ip], "tid": tid,
                           "ph": "M", "args": {"name": comp}})

        for t in filtered:
            pd = t["profileData"]
            ip = pd["nodeIpAddress"]
            comp = f"{pd.get('componentType', '')}:{pd.get('componentId', '')}"
            pid = node_ip_to_pid[ip]
            tid = tid_map[(ip, comp)]
            for ev in pd["events"]:
                start_us = ev["startTime"] / 1000.0  # ns → µs
                dur_us = (ev["endTime"] - ev["startTime"]) / 1000.0
                extra = {}
                if ev.get("extraData"):
                    try:
                        extra = json.loads(ev["extraData"])
                    except ValueError:
                        extra = {}
                task_id_for_args = extra.get("task_id") or t["taskId"]
                func = t.get("funcOrClassName") or ""
                args = {
                    "task_id": task_id_for_args,
                    "job_id": t.get("jobId"),
                    "attempt_number": t.get("taskAttempt", 0),
                    "func_or_class_name": func,
                    "actor_id": extract_actor_id_from_task_id(
                        task_id_for_args) or None,
                }
                name = ev["eventName"]
                display = name
                if name.startswith("task::") and extra.get("name"):
                    display = extra["name"]
                    args["name"] = extra["name"]
                events.append({
                    "cat": name, "name": display, "pid": pid, "tid": tid,
                    "ts": start_us, "dur": dur_us, "ph": "X",
                    "cname": chrome_trace_color(name), "args": args})
        return events


_COLOR_MAP = {
    "task:deserialize_arguments": "rail_load",
    "task:execute": "rail_animation",
    "task:store_outputs": "rail_idle",
    "task:submit_task": "rail_response",
    "task": "rail_response",
    "worker_idle": "cq_build_abandoned",
    "ray.get": "good",
    "ray.put": "terrible",
    "ray.wait": "vsync_highlight_color",
    "submit_task": "background_memory_dump",
    "wait_for_function": "detailed_memory_dump",
    "fetch_and_run_function": "detailed_memory_dump",
    "register_remote_function": "detailed_memory_dump",
}


def chrome_trace_color(event_name: str) -> str:
    """timeline.go getChromeTraceColor (Ray profiling.py color mapping)."""
    if event_name.startswith("task::"):
        return "generic_work"
    return _COLOR_MAP.get(event_name, "generic_work")


def decode_event_file_bytes(file_name: str, raw: bytes) -> List[Dict[str, Any]]:
    """DecodeEventFileBytes (eventserver.go:67-103): JSON array (legacy) vs
    JSONL auto-detection; malformed JSONL lines are skipped, not fatal."""
    text = raw.decode(errors="replace").lstrip()
    if not text:
        return []
    if text[0] == "[":
        try:
            out = json.loads(text)
            return out if isinstance(out, list) else []
        except ValueError as e:
            raise ValueError(f"unmarshal JSON array {file_name}: {e}")
    out = []
    for line in text.splitlines():
        line = line.strip()
        if not line:
            continue
        try:
            obj = json.loads(line)
        except ValueError:
            logger.warning("skipping malformed JSONL line in %s", file_name)
            continue
        if isinstance(obj, dict):
            out.append(obj)
    return out


def load_session(storage: StorageReader, prefix: str) -> SessionState:
    """Replay every stored event batch under {prefix}/events/ plus the
    dashboard log events under {prefix}/logs/*/events/ (ProcessSingleSession
    + LogEventReader.ReadLogEvents analogs)."""
    state = SessionState()
    for path in storage.list(f"{prefix}/events"):
        try:
            raw = storage.read(path)
            if path.endswith(".gz"):
                raw = decompress(raw)
        except Exception:
            # one corrupt batch must not kill post-mortem browsing of the
            # rest of the session
            logger.warning("skipping unreadable event batch %s", path)
            continue
        try:
            events = decode_event_file_bytes(path, raw)
        except ValueError:
            logger.warning("skipping undecodable event batch %s", path)
            continue
        for event in events:
            state.apply(event)
    # dashboard log events: {prefix}/logs/{node}/events/event_*.log[.gz]
    for path in storage.list(f"{prefix}/logs"):
        name = path.rsplit("/", 1)[-1]
        if "/events/" not in path or not name.startswith("event_"):
            continue
        try:
            raw = storage.read(path)
            if path.endswith(".gz"):
                raw = decompress(raw)
        except Exception:
            logger.warning("skipping unreadable log-event file %s", path)
            continue
        for line in raw.decode(errors="replace").splitlines():
            line = line.strip()
            if not line:
                continue
            try:
                state.apply_log_event(json.loads(line))
            except ValueError:
                continue
    return state
