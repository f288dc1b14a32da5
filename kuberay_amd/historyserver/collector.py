"""History collector (reference: historyserver/pkg/collector/ + the sidecar
injection in common/pod.go:304-315).

Polls a Ray node's event/log endpoints and writes gzip-compressed batches
into storage under ``{cluster_ns}/{cluster_name}/{session}/...``. Runs as a
sidecar in head/worker pods (``build_collector_container``) or in-process
for tests.
"""
from __future__ import annotations

import json
import threading
import time
from typing import Any, Callable, Dict, List, Optional

from ..kube.objects import Container, EnvVar, EnvVarSource, ResourceRequirements
from ..utils import constants as C
from .storage import StorageWriter, compress

DEFAULT_EVENTS_PORT = 8084
DEFAULT_PUSH_INTERVAL_S = 10.0


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


def main() -> int:  # sidecar entrypoint
    import os

    import httpx

    from .storage import storage_for

    storage = storage_for(os.environ.get("STORAGE_BACKEND", "local"),
                          root=os.environ.get("STORAGE_ROOT_DIR",
                                              "/var/lib/kuberay-history"))
    pod_ip = os.environ.get("POD_IP", "127.0.0.1")
    port = int(os.environ.get("EVENTS_PORT", DEFAULT_EVENTS_PORT))
    http = httpx.Client(timeout=5.0)

    def fetch_events():
        try:
            resp = http.get(f"http://{pod_ip}:{port}/v1/events")
            if resp.status_code == 200:
                data = resp.json()
                return data if isinstance(data, list) else data.get("events", [])
        except httpx.HTTPError:
            pass
        return []

    collector = Collector(
        storage,
        cluster_name=os.environ.get("OWNER_NAME", "unknown"),
        namespace=os.environ.get(C.RAY_CLUSTER_NAMESPACE, "default"),
        fetch_events=fetch_events,
        push_interval_s=float(os.environ.get("PUSH_INTERVAL",
                                             DEFAULT_PUSH_INTERVAL_S)))
    collector.start()
    try:
        while True:
            time.sleep(60)
    except KeyboardInterrupt:
        collector.stop()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
