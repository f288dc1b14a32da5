"""History server HTTP API (reference: historyserver/pkg/historyserver/
{server,router,session_loader,timeline}.go — a Ray-dashboard-compatible
read-only API over stored sessions)."""
from __future__ import annotations

import json
from typing import Dict, List, Optional

from fastapi import FastAPI, HTTPException

from .eventserver import SessionState, load_session
from .storage import StorageReader, storage_for


class SessionLoader:
    """Lazily loads + caches replayed sessions (session_loader.go)."""

    def __init__(self, storage: StorageReader):
        self.storage = storage
        self._cache: Dict[str, SessionState] = {}

    def sessions(self):
        seen = set()
        for path in self.storage.list(""):
            parts = path.split("/")
            if len(parts) >= 3:
                seen.add("/".join(parts[:3]))  # ns/cluster/session
        return sorted(seen)

    def load(self, namespace: str, cluster: str, session: str) -> SessionState:
        prefix = f"{namespace}/{cluster}/{session}"
        if prefix not in self._cache:
            self._cache[prefix] = load_session(self.storage, prefix)
        return self._cache[prefix]


def create_history_app(storage: Optional[StorageReader] = None) -> FastAPI:
    storage = storage or storage_for("local")
    loader = SessionLoader(storage)
    app = FastAPI(title="kuberay-amd-history-server")
    app.state.loader = loader

    @app.get("/api/sessions")
    def sessions():
        return {"sessions": loader.sessions()}

    def _state(ns: str, cluster: str, session: str) -> SessionState:
        return loader.load(ns, cluster, session)

    def _list_options(filter_keys, filter_predicates, filter_values,
                      limit) -> tuple:
        """Ray state-API list options (router.go:1814-1896 ParseOptionsFromReq
        analog): parallel filter_keys/filter_predicates/filter_values triples
        with '=' / '!=' predicates, plus a result limit."""
        keys = filter_keys or []
        preds = filter_predicates or ["="] * len(keys)
        vals = filter_values or []
        if not (len(keys) == len(vals) and len(preds) == len(keys)):
            raise HTTPException(
                400, "filter_keys, filter_predicates and filter_values "
                     "must have equal lengths")
        for p in preds:
            if p not in ("=", "!="):
                raise HTTPException(400, f"unsupported predicate {p!r}")
        return list(zip(keys, preds, vals)), limit

    def _apply_filters(items, filters, limit):
        """Returns (page, num_after_truncation, num_filtered)."""
        total = len(items)
        if filters:
            def keep(it):
                for key, pred, val in filters:
                    have = str(it.get(key, ""))
                    if (pred == "=") != (have == val):
                        return False
                return True
            items = [it for it in items if keep(it)]
        filtered = len(items)
        return items[:limit], total, filtered

    def _envelope(key, page, total, filtered):
        return {"result": True, "msg": f"{key} fetched.",
                "data": {key: page,
                         "num_after_truncation": total,
                         "num_filtered": filtered}}

    def _listing(key, items, filter_keys, filter_predicates, filter_values,
                 limit):
        filters, limit = _list_options(filter_keys, filter_predicates,
                                       filter_values, limit)
        page, total, filtered = _apply_filters(items, filters, limit)
        return _envelope(key, page, total, filtered)

    from fastapi import Query

    _FK = Query(default=None, alias="filter_keys")
    _FP = Query(default=None, alias="filter_predicates")
    _FV = Query(default=None, alias="filter_values")

    @app.get("/api/sessions/{ns}/{cluster}/{session}/jobs")
    def jobs(ns: str, cluster: str, session: str,
             filter_keys: Optional[List[str]] = _FK,
             filter_predicates: Optional[List[str]] = _FP,
             filter_values: Optional[List[str]] = _FV,
             limit: int = 10000):
        return _listing("jobs", list(_state(ns, cluster, session).jobs.values()),
                        filter_keys, filter_predicates, filter_values, limit)

    @app.get("/api/sessions/{ns}/{cluster}/{session}/jobs/{job_id}")
    def job_detail(ns: str, cluster: str, session: str, job_id: str):
        job = _state(ns, cluster, session).jobs.get(job_id)
        if job is None:
            raise HTTPException(404, f"job {job_id} not found")
        return {"result": True, "data": {"detail": job}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/tasks")
    def tasks(ns: str, cluster: str, session: str,
              filter_keys: Optional[List[str]] = _FK,
              filter_predicates: Optional[List[str]] = _FP,
              filter_values: Optional[List[str]] = _FV,
              limit: int = 10000):
        return _listing("tasks",
                        list(_state(ns, cluster, session).tasks.values()),
                        filter_keys, filter_predicates, filter_values, limit)

    @app.get("/api/sessions/{ns}/{cluster}/{session}/tasks/summarize")
    def tasks_summarize(ns: str, cluster: str, session: str):
        """router.go:184-190 — per-function aggregate of task states."""
        summary: Dict[str, Dict] = {}
        for t in _state(ns, cluster, session).tasks.values():
            name = t.get("funcOrClassName") or t.get("name") or "unknown"
            entry = summary.setdefault(name, {
                "func_or_class_name": name, "type": t.get("taskType"),
                "state_counts": {}})
            state = t.get("state") or "UNKNOWN"
            entry["state_counts"][state] = \
                entry["state_counts"].get(state, 0) + 1
        return {"result": True, "data": {
            "summary": sorted(summary.values(),
                              key=lambda e: e["func_or_class_name"]),
            "total_tasks": len(_state(ns, cluster, session).tasks)}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/tasks/timeline")
    def tasks_timeline(ns: str, cluster: str, session: str,
                       job_id: Optional[str] = None):
        """Ray /api/v0/tasks/timeline analog (timeline.go:13)."""
        return _state(ns, cluster, session).timeline(job_id)

    @app.get("/api/sessions/{ns}/{cluster}/{session}/tasks/{task_id}")
    def task_detail(ns: str, cluster: str, session: str, task_id: str):
        """Latest attempt for the task, with every attempt inline (the
        reference lists one row per attempt; detail groups them)."""
        state = _state(ns, cluster, session)
        attempts = sorted(
            (t for t in state.tasks.values() if t.get("taskId") == task_id),
            key=lambda t: t.get("taskAttempt", 0))
        if not attempts:
            raise HTTPException(404, f"task {task_id} not found")
        detail = dict(attempts[-1])
        detail["attempts"] = attempts
        return {"result": True, "data": {"detail": detail}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/events")
    def log_events(ns: str, cluster: str, session: str,
                   job_id: Optional[str] = None):
        """Dashboard /events analog (log_event_reader.go): log events
        grouped by job."""
        ev = _state(ns, cluster, session).log_events
        if job_id is not None:
            return {"result": True,
                    "data": {"events": {job_id: ev.get(job_id, [])}}}
        return {"result": True, "data": {"events": ev}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/actors")
    def actors(ns: str, cluster: str, session: str,
               filter_keys: Optional[List[str]] = _FK,
               filter_predicates: Optional[List[str]] = _FP,
               filter_values: Optional[List[str]] = _FV,
               limit: int = 10000):
        return _listing("actors",
                        list(_state(ns, cluster, session).actors.values()),
                        filter_keys, filter_predicates, filter_values, limit)

    @app.get("/api/sessions/{ns}/{cluster}/{session}/nodes")
    def nodes(ns: str, cluster: str, session: str):
        return {"data": {"nodes": list(_state(ns, cluster, session).nodes.values())}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/nodes/{node_id}")
    def node_detail(ns: str, cluster: str, session: str, node_id: str):
        node = _state(ns, cluster, session).nodes.get(node_id)
        if node is None:
            raise HTTPException(404, f"node {node_id} not found")
        return {"result": True, "data": {"detail": node}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/actors/{actor_id}")
    def actor_detail(ns: str, cluster: str, session: str, actor_id: str):
        actor = _state(ns, cluster, session).actors.get(actor_id)
        if actor is None:
            raise HTTPException(404, f"actor {actor_id} not found")
        return {"result": True, "data": {"detail": actor}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/cluster_status")
    def cluster_status(ns: str, cluster: str, session: str):
        """cluster_status.go analog: autoscaler-style summary built from the
        replayed node/task/actor state."""
        state = _state(ns, cluster, session)
        alive, failed = [], []
        for node in state.nodes.values():
            (failed if (node.get("state") or "").upper() == "DEAD"
             else alive).append(node)
        demands: Dict[str, Dict] = {}
        pending_states = {"", "PENDING", "PENDING_NODE_ASSIGNMENT",
                          "PENDING_ARGS_AVAIL", "PENDING_CREATION",
                          "DEPENDENCIES_UNREADY"}
        for item in list(state.tasks.values()) + list(state.actors.values()):
            if (item.get("state") or "").upper() not in pending_states:
                continue
            res = item.get("requiredResources") or {"CPU": 1}
            key = json.dumps(res, sort_keys=True)
            entry = demands.setdefault(key, {"resources": res, "count": 0})
            entry["count"] += 1
        lines = ["======== Cluster status (replayed) ========",
                 f"Active: {len(alive)} node(s)"]
        for node in failed[:20]:
            lines.append(
                f"Failed: {node.get('nodeIpAddress') or node['nodeId']}")
        if demands:
            lines.append("Pending demands:")
            for d in demands.values():
                lines.append(f"  {d['resources']}: {d['count']}+")
        return {"result": True, "data": {"clusterStatus": {
            "activeNodes": len(alive),
            "failedNodes": [n["nodeId"] for n in failed],
            "pendingDemands": list(demands.values()),
            "text": "\n".join(lines)}}}

    @app.get("/api/grafana_health")
    def grafana_health():
        """router.go:124 analog — no Grafana bundled in this offline image."""
        return {"result": False, "msg": "grafana is not configured"}

    @app.get("/api/prometheus_health")
    def prometheus_health():
        return {"result": False, "msg": "prometheus is not configured"}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/timeline")
    def timeline(ns: str, cluster: str, session: str):
        return _state(ns, cluster, session).timeline()

    @app.get("/api/sessions/{ns}/{cluster}/{session}/logs")
    def list_logs(ns: str, cluster: str, session: str):
        """router.go:144-149 getNodeLogs analog: enumerate stored log files."""
        prefix = f"{ns}/{cluster}/{session}/logs/"
        names = sorted(p[len(prefix):].removesuffix(".gz")
                       for p in storage.list(prefix))
        return {"result": True, "data": {"logs": names}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/logs/{log_name}")
    def logs(ns: str, cluster: str, session: str, log_name: str,
             lines: int = 0, offset: int = 0):
        """``lines`` > 0 tails that many lines (Ray /api/v0/logs semantics);
        ``offset`` skips leading lines first — together they paginate."""
        from .storage import decompress
        path = f"{ns}/{cluster}/{session}/logs/{log_name}.gz"
        if not storage.exists(path):
            raise HTTPException(404, f"log {log_name} not found")
        text = decompress(storage.read(path)).decode()
        total = text.count("\n") + (0 if text.endswith("\n") or not text
                                    else 1)
        if offset or lines:
            split = text.splitlines(keepends=True)
            split = split[offset:]
            if lines > 0:
                split = split[-lines:] if offset == 0 else split[:lines]
            text = "".join(split)
        return {"logs": text, "total_lines": total}

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    return app


def main(argv=None) -> int:
    import argparse

    import uvicorn

    parser = argparse.ArgumentParser(prog="kuberay-amd-history-server")
    parser.add_argument("--port", type=int, default=8089)
    parser.add_argument("--storage-backend", default="local")
    parser.add_argument("--storage-root", default="/var/lib/kuberay-history")
    args = parser.parse_args(argv)
    storage = storage_for(args.storage_backend, root=args.storage_root)
    uvicorn.run(create_history_app(storage), host="0.0.0.0", port=args.port)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
