"""History collector (reference: historyserver/pkg/collector/eventcollector/
eventcollector.go:31-1152 + the sidecar injection in common/pod.go:304-315).

Two modes:

* ``EventCollector`` — the reference design: a disk-first HTTP receiver.
  Ray's export-event pipeline POSTs batches to ``/v1/events``; events are
  appended to local JSONL files per category (node events vs per-job),
  rotated by size/age, gzip-compressed, and uploaded to storage under the
  cluster-session prefix. Files left on disk by a crash are resumed on the
  next start (eventcollector.go resumePendingFiles :885).
* ``Collector`` — a poll-compress-write loop kept for environments where
  the collector must pull from the dashboard agent instead.

Runs as a sidecar in head/worker pods (``build_collector_container``) or
in-process for tests.
"""
from __future__ import annotations

import json
import logging
import os
import threading
import time
from typing import Any, Callable, Dict, List, Optional

from ..kube.objects import Container, EnvVar, EnvVarSource, ResourceRequirements
from ..utils import constants as C
from .storage import StorageWriter, compress

logger = logging.getLogger("kuberay.historyserver.collector")

DEFAULT_EVENTS_PORT = 8084
DEFAULT_PUSH_INTERVAL_S = 10.0
DEFAULT_MAX_FILE_BYTES = 16 * 1024 * 1024
DEFAULT_ROTATION_INTERVAL_S = 60.0
DEFAULT_MAX_DISK_BYTES = 1024 * 1024 * 1024

# event types that describe the node, not a job (eventcollector.go:31-50)
_NODE_EVENT_TYPES = {"NODE_DEFINITION_EVENT", "NODE_LIFECYCLE_EVENT"}
_EVENT_TYPES_WITH_JOB_ID = (
    "taskDefinitionEvent", "taskLifecycleEvent", "actorTaskDefinitionEvent",
    "actorDefinitionEvent", "driverJobDefinitionEvent",
    "driverJobLifecycleEvent", "taskProfileEvents")
_NODE_CATEGORY = "node-events"


def _is_safe_path_component(s: str) -> bool:
    return bool(s) and "/" not in s and "\\" not in s and ".." not in s


def _job_id_of(event: Dict[str, Any]) -> str:
    """getJobID (eventcollector.go:1027-1046): jobId from any known nested
    payload, normalized to hex for path safety."""
    from .eventserver import normalize_id
    for field in _EVENT_TYPES_WITH_JOB_ID:
        nested = event.get(field)
        if isinstance(nested, dict) and nested.get("jobId"):
            job_id = normalize_id(str(nested["jobId"]))
            if not _is_safe_path_component(job_id):
                logger.warning("ignoring unsafe jobId %r; filing under %s",
                               job_id, _NODE_CATEGORY)
                return ""
            return job_id
    return ""


def categorize(event: Dict[str, Any]) -> str:
    """eventcollector.go categorize (:453-465): node events (and events
    without a job) file under the node category; the rest under job/{id}."""
    if event.get("eventType") in _NODE_EVENT_TYPES:
        return _NODE_CATEGORY
    job_id = _job_id_of(event)
    return f"job/{job_id}" if job_id else _NODE_CATEGORY


class EventCollector:
    """Disk-first event receiver with rotation + async upload."""

    def __init__(self, storage: StorageWriter, cluster_name: str,
                 namespace: str = "default", session: str = "session-1",
                 node_id: str = "node", data_dir: Optional[str] = None,
                 max_file_bytes: int = DEFAULT_MAX_FILE_BYTES,
                 rotation_interval_s: float = DEFAULT_ROTATION_INTERVAL_S,
                 compression: bool = True,
                 max_disk_bytes: int = DEFAULT_MAX_DISK_BYTES):
        import tempfile
        self.storage = storage
        self.cluster_name = cluster_name
        self.namespace = namespace
        self.session = session
        self.node_id = node_id
        self.data_dir = data_dir or tempfile.mkdtemp(prefix="kuberay-events-")
        self.max_file_bytes = max_file_bytes
        self.rotation_interval_s = rotation_interval_s
        self.compression = compression
        self.max_disk_bytes = max_disk_bytes
        self._lock = threading.Lock()
        # category -> {"path": str, "size": int, "opened": float}
        self._active: Dict[str, Dict[str, Any]] = {}
        self._disk_used = 0
        self._draining = False
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.events_received = 0
        self.events_dropped = 0

    @property
    def prefix(self) -> str:
        return f"{self.namespace}/{self.cluster_name}/{self.session}"

    # -- ingest (PersistEvents :293-451) --------------------------------
    def persist_events(self, events: List[Dict[str, Any]]) -> int:
        """Append a batch to local JSONL files. Returns events accepted."""
        if self._draining:
            raise RuntimeError("event collector is shutting down")
        accepted = 0
        with self._lock:
            for event in events:
                if not isinstance(event, dict):
                    continue
                if self._disk_used >= self.max_disk_bytes:
                    # disk pressure: drop rather than fill the node disk
                    # (eventcollector.go underDiskPressure :1004)
                    self.events_dropped += 1
                    continue
                line = (json.dumps(event, sort_keys=True) + "\n").encode()
                state = self._active_file_locked(categorize(event))
                with open(state["path"], "ab") as f:
                    f.write(line)
                state["size"] += len(line)
                self._disk_used += len(line)
                accepted += 1
                self.events_received += 1
                if state["size"] >= self.max_file_bytes:
                    self._rotate_locked(state["category"])
        return accepted

    def _active_file_locked(self, category: str) -> Dict[str, Any]:
        state = self._active.get(category)
        if state is None:
            safe = category.replace("/", "_")
            path = os.path.join(
                self.data_dir, f"active-{safe}-{int(time.time()*1e9)}.jsonl")
            state = {"path": path, "size": 0, "opened": time.time(),
                     "category": category}
            open(path, "ab").close()
            self._active[category] = state
        return state

    # -- rotation + upload (:550-814) ------------------------------------
    def _rotate_locked(self, category: str) -> None:
        state = self._active.pop(category, None)
        if state is None or state["size"] == 0:
            if state is not None:
                try:
                    os.remove(state["path"])
                except OSError:
                    pass
            return
        pending = state["path"].replace("active-", "pending-")
        os.replace(state["path"], pending)
        self._upload(pending, category, state["opened"])

    def _storage_key(self, category: str, opened: float) -> str:
        """buildEventStorageKey (:816-848): {node}-{YYYY-MM-DD-HH}-{ns}."""
        hour = time.strftime("%Y-%m-%d-%H", time.gmtime(opened))
        ext = ".jsonl.gz" if self.compression else ".jsonl"
        name = f"{self.node_id}-{hour}-{int(opened*1e9)}{ext}"
        return f"{self.prefix}/events/{category}/{name}"

    def _upload(self, path: str, category: str, opened: float) -> None:
        try:
            with open(path, "rb") as f:
                data = f.read()
            payload = compress(data) if self.compression else data
            self.storage.write(self._storage_key(category, opened), payload)
            self._disk_used = max(0, self._disk_used - len(data))
            os.remove(path)
        except Exception:
            # upload failed: the pending file stays on disk for
            # resume_pending() after the next start (crash-safe)
            logger.warning("upload of %s failed; left on disk for resume",
                           path, exc_info=True)

    def rotate_all(self) -> None:
        with self._lock:
            for category in list(self._active):
                self._rotate_locked(category)

    def resume_pending(self) -> None:
        """resumePendingFiles (:885-1002): upload files a crash left behind."""
        for name in sorted(os.listdir(self.data_dir)):
            if not name.startswith(("pending-", "active-")):
                continue
            path = os.path.join(self.data_dir, name)
            # category was flattened into the file name; recover best-effort
            middle = name.split("-", 1)[1].rsplit("-", 1)[0]
            category = middle.replace("_", "/", 1) if \
                middle.startswith("job_") else _NODE_CATEGORY
            self._upload(path, category, os.path.getmtime(path))

    # -- lifecycle --------------------------------------------------------
    def start(self) -> None:
        self.resume_pending()

        def loop():
            while not self._stop.wait(self.rotation_interval_s):
                with self._lock:
                    now = time.time()
                    for category, state in list(self._active.items()):
                        if now - state["opened"] >= self.rotation_interval_s \
                                and state["size"] > 0:
                            self._rotate_locked(category)
        self._thread = threading.Thread(target=loop, name="event-rotation",
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        """Drain: reject new events, rotate + upload everything (:266-291)."""
        self._draining = True
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=3)
        self.rotate_all()


def build_collector_container(collector_options, node_type: str,
                              cluster_name: str, namespace: str,
                              fqdn_ray_ip: str) -> Container:
    """common/pod.go BuildCollectorContainer analog."""
    opts = collector_options
    container = Container(
        name="history-collector",
        image=(opts.image if opts and opts.image else None),
        image_pull_policy=(opts.image_pull_policy if opts else None) or "IfNotPresent",
        command=["python", "-m", "kuberay_amd.historyserver.collector"],
        env=[
            EnvVar(name="POD_IP", value_from=EnvVarSource(
                field_ref={"fieldPath": "status.podIP"})),
            EnvVar(name="RAY_ROLE", value=node_type),
            EnvVar(name="OWNER_KIND", value="RayCluster"),
            EnvVar(name="OWNER_NAME", value=cluster_name),
            EnvVar(name=C.RAY_CLUSTER_NAMESPACE, value=namespace),
            EnvVar(name="EVENTS_PORT", value=str(DEFAULT_EVENTS_PORT)),
            EnvVar(name="STORAGE_BACKEND", value="local"),
            EnvVar(name="STORAGE_ROOT_DIR", value="/var/lib/kuberay-history"),
            EnvVar(name="PUSH_INTERVAL", value=str(int(DEFAULT_PUSH_INTERVAL_S))),
        ],
        resources=(opts.resources if opts and opts.resources else
                   ResourceRequirements(limits={"cpu": "200m", "memory": "256Mi"},
                                        requests={"cpu": "100m", "memory": "128Mi"})),
    )
    if opts and opts.env:
        for e in opts.env:
            container.set_env_if_absent(e.name, e.value or "")
    return container


class Collector:
    """Poll-compress-write loop. ``fetch_events`` is injectable (tests use a
    fake; the sidecar uses the Ray dashboard-agent HTTP endpoint)."""

    def __init__(self, storage: StorageWriter, cluster_name: str,
                 namespace: str = "default", session: str = "session-1",
                 fetch_events: Optional[Callable[[], List[Dict[str, Any]]]] = None,
                 fetch_logs: Optional[Callable[[], Dict[str, str]]] = None,
                 push_interval_s: float = DEFAULT_PUSH_INTERVAL_S):
        self.storage = storage
        self.cluster_name = cluster_name
        self.namespace = namespace
        self.session = session
        self.fetch_events = fetch_events or (lambda: [])
        self.fetch_logs = fetch_logs or (lambda: {})
        self.push_interval_s = push_interval_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._batch_idx = 0

    @property
    def prefix(self) -> str:
        return f"{self.namespace}/{self.cluster_name}/{self.session}"

    def push_once(self) -> int:
        """One poll+write cycle; returns number of events written."""
        events = self.fetch_events()
        if events:
            payload = "\n".join(json.dumps(e, sort_keys=True) for e in events)
            path = f"{self.prefix}/events/batch-{self._batch_idx:06d}.jsonl.gz"
            self.storage.write(path, compress(payload.encode()))
            self._batch_idx += 1
        for log_name, content in (self.fetch_logs() or {}).items():
            self.storage.write(f"{self.prefix}/logs/{log_name}.gz",
                               compress(content.encode()))
        return len(events)

    def start(self) -> None:
        def loop():
            while not self._stop.is_set():
                try:
                    self.push_once()
                except Exception:
                    pass
                self._stop.wait(self.push_interval_s)
        self._thread = threading.Thread(target=loop, name="history-collector",
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
        self.push_once()


def create_receiver_app(collector: EventCollector):
    """The /v1/events HTTP receiver Ray's export pipeline POSTs to
    (eventcollector.go Run :238-247)."""
    # module-global import: `from __future__ import annotations` stringifies
    # the endpoint's type hints and FastAPI resolves them against module
    # globals — function-local imports would make Request/Response
    # unresolvable and silently degrade to query params
    global Request, Response
    from fastapi import FastAPI, Request, Response

    app = FastAPI(title="kuberay-amd-event-collector")

    @app.post("/v1/events")
    async def persist(request: Request, response: Response):
        try:
            body = await request.json()
        except Exception:
            response.status_code = 400
            return {"error": "invalid JSON"}
        events = body if isinstance(body, list) else body.get("events", [])
        try:
            accepted = collector.persist_events(events)
        except RuntimeError:
            response.status_code = 503
            response.headers["Retry-After"] = "5"
            return {"error": "event collector is shutting down"}
        return {"accepted": accepted}

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    return app


def main() -> int:  # sidecar entrypoint
    import uvicorn

    from .storage import storage_for

    storage = storage_for(os.environ.get("STORAGE_BACKEND", "local"),
                          root=os.environ.get("STORAGE_ROOT_DIR",
                                              "/var/lib/kuberay-history"))
    port = int(os.environ.get("EVENTS_PORT", DEFAULT_EVENTS_PORT))
    collector = EventCollector(
        storage,
        cluster_name=os.environ.get("OWNER_NAME", "unknown"),
        namespace=os.environ.get(C.RAY_CLUSTER_NAMESPACE, "default"),
        node_id=os.environ.get("POD_IP", "node").replace(".", "-"),
        rotation_interval_s=float(os.environ.get(
            "PUSH_INTERVAL", DEFAULT_PUSH_INTERVAL_S)))
    collector.start()
    try:
        uvicorn.run(create_receiver_app(collector), host="0.0.0.0", port=port)
    finally:
        collector.stop()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
