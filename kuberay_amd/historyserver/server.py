"""History server HTTP API (reference: historyserver/pkg/historyserver/
{server,router,session_loader,timeline}.go — a Ray-dashboard-compatible
read-only API over stored sessions)."""
from __future__ import annotations

from typing import Dict, Optional

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

    @app.get("/api/sessions/{ns}/{cluster}/{session}/jobs")
    def jobs(ns: str, cluster: str, session: str):
        return {"data": {"jobs": list(_state(ns, cluster, session).jobs.values())}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/tasks")
    def tasks(ns: str, cluster: str, session: str):
        return {"data": {"tasks": list(_state(ns, cluster, session).tasks.values())}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/actors")
    def actors(ns: str, cluster: str, session: str):
        return {"data": {"actors": list(_state(ns, cluster, session).actors.values())}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/nodes")
    def nodes(ns: str, cluster: str, session: str):
        return {"data": {"nodes": list(_state(ns, cluster, session).nodes.values())}}

    @app.get("/api/sessions/{ns}/{cluster}/{session}/timeline")
    def timeline(ns: str, cluster: str, session: str):
        return _state(ns, cluster, session).timeline()

    @app.get("/api/sessions/{ns}/{cluster}/{session}/logs/{log_name}")
    def logs(ns: str, cluster: str, session: str, log_name: str):
        from .storage import decompress
        path = f"{ns}/{cluster}/{session}/logs/{log_name}.gz"
        if not storage.exists(path):
            raise HTTPException(404, f"log {log_name} not found")
        return {"logs": decompress(storage.read(path)).decode()}

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
