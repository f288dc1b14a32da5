"""Kubernetes REST façade over the in-memory API server.

Speaks enough of the kube-apiserver HTTP surface (typed paths, status
subresource, labelSelector, chunked ``?watch=true`` streams) for the
deployable REST stack (kuberay_amd/kube/rest.py RestClient +
RestApiServerAdapter) to run against it unchanged — the integration-test
analog of envtest's real apiserver, and a usable standalone control-plane
endpoint (``python -m kuberay_amd.kube.httpserver``).
"""
from __future__ import annotations

import json
import re
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional, Tuple
from urllib.parse import parse_qs, urlparse

from .rest import RESOURCES
from .store import AlreadyExistsError, ApiError, ConflictError, InMemoryApiServer, NotFoundError

# path prefix -> kind lookup
_KIND_BY_PREFIX_PLURAL = {(prefix, plural): kind
                          for kind, (prefix, plural) in RESOURCES.items()}

_PATH_RE = re.compile(
    r"^(?P<prefix>/api/v1|/apis/[^/]+/[^/]+)"
    r"(?:/namespaces/(?P<ns>[^/]+))?"
    r"/(?P<plural>[^/?]+)"
    r"(?:/(?P<name>[^/?]+))?"
    r"(?:/(?P<sub>status))?$")


def _parse_path(path: str) -> Optional[Tuple[str, Optional[str], Optional[str], Optional[str]]]:
    m = _PATH_RE.match(path)
    if not m:
        return None
    kind = _KIND_BY_PREFIX_PLURAL.get((m.group("prefix"), m.group("plural")))
    if kind is None:
        return None
    name, sub = m.group("name"), m.group("sub")
    if name == "status":  # /plural/name/status matched oddly
        name, sub = None, "status"
    return kind, m.group("ns"), name, sub


class _Handler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"
    server_version = "kuberay-amd-kubeapi/1.0"

    @property
    def store(self) -> InMemoryApiServer:
        return self.server.store  # type: ignore[attr-defined]

    def log_message(self, *a):
        pass

    def _send_json(self, code: int, obj) -> None:
        body = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _send_error_status(self, e: ApiError) -> None:
        reason = {404: "NotFound", 409: "Conflict"}.get(e.code, "BadRequest")
        if isinstance(e, AlreadyExistsError):
            reason = "AlreadyExists"
        self._send_json(e.code, {
            "kind": "Status", "apiVersion": "v1", "status": "Failure",
            "reason": reason, "message": f"{reason}: {e.message}",
            "code": e.code})

    def _body(self):
        length = int(self.headers.get("Content-Length", 0))
        return json.loads(self.rfile.read(length)) if length else {}

    # ------------------------------------------------------------------
    def do_GET(self):  # noqa: N802
        url = urlparse(self.path)
        parsed = _parse_path(url.path)
        if parsed is None:
            return self._send_json(404, {"message": f"unknown path {url.path}"})
        kind, ns, name, _ = parsed
        params = parse_qs(url.query)
        if name:
            obj = self.store.try_get(kind, ns or "default", name)
            if obj is None:
                return self._send_error_status(NotFoundError(f"{kind} {name}"))
            return self._send_json(200, obj)
        if params.get("watch", ["false"])[0] == "true":
            return self._stream_watch(kind)
        selector = None
        if "labelSelector" in params:
            selector = dict(kv.split("=", 1)
                            for kv in params["labelSelector"][0].split(","))
        items = self.store.list(kind, ns, selector)
        self._send_json(200, {"kind": f"{kind}List", "apiVersion": "v1",
                              "items": items})

    def _stream_watch(self, kind: str) -> None:
        watcher = self.store.watch({kind})
        try:
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Transfer-Encoding", "chunked")
            self.end_headers()
            while not self.server.stopping:  # type: ignore[attr-defined]
                ev = watcher.next(timeout=0.5)
                if ev is None:
                    continue
                event_type, obj = ev
                line = json.dumps({"type": event_type, "object": obj}) + "\n"
                data = line.encode()
                self.wfile.write(f"{len(data):x}\r\n".encode() + data + b"\r\n")
                self.wfile.flush()
            self.wfile.write(b"0\r\n\r\n")
        except (BrokenPipeError, ConnectionResetError, OSError):
            pass
        finally:
            watcher.stop()

    def do_POST(self):  # noqa: N802
        parsed = _parse_path(urlparse(self.path).path)
        if parsed is None:
            return self._send_json(404, {"message": "unknown path"})
        kind, ns, _, _ = parsed
        obj = self._body()
        obj.setdefault("kind", kind)
        obj.setdefault("metadata", {}).setdefault("namespace", ns or "default")
        try:
            self._send_json(201, self.store.create(obj))
        except ApiError as e:
            self._send_error_status(e)

    def do_PUT(self):  # noqa: N802
        parsed = _parse_path(urlparse(self.path).path)
        if parsed is None:
            return self._send_json(404, {"message": "unknown path"})
        kind, ns, name, sub = parsed
        obj = self._body()
        obj.setdefault("kind", kind)
        obj.setdefault("metadata", {}).setdefault("namespace", ns or "default")
        if name:
            obj["metadata"]["name"] = name
        try:
            self._send_json(200, self.store.update(obj, subresource=sub))
        except ApiError as e:
            self._send_error_status(e)

    def do_PATCH(self):  # noqa: N802
        parsed = _parse_path(urlparse(self.path).path)
        if parsed is None:
            return self._send_json(404, {"message": "unknown path"})
        kind, ns, name, sub = parsed
        try:
            self._send_json(200, self.store.patch_merge(
                kind, ns or "default", name, self._body(), subresource=sub))
        except ApiError as e:
            self._send_error_status(e)

    def do_DELETE(self):  # noqa: N802
        parsed = _parse_path(urlparse(self.path).path)
        if parsed is None:
            return self._send_json(404, {"message": "unknown path"})
        kind, ns, name, _ = parsed
        try:
            self.store.delete(kind, ns or "default", name)
            self._send_json(200, {"kind": "Status", "status": "Success"})
        except ApiError as e:
            self._send_error_status(e)


class KubeApiFacade:
    """Threaded HTTP server over an InMemoryApiServer."""

    def __init__(self, store: Optional[InMemoryApiServer] = None,
                 host: str = "127.0.0.1", port: int = 0):
        self.store = store or InMemoryApiServer()
        self._httpd = ThreadingHTTPServer((host, port), _Handler)
        self._httpd.store = self.store  # type: ignore[attr-defined]
        self._httpd.stopping = False  # type: ignore[attr-defined]
        self._httpd.daemon_threads = True
        self._thread: Optional[threading.Thread] = None

    @property
    def url(self) -> str:
        host, port = self._httpd.server_address[:2]
        return f"http://{host}:{port}"

    def start(self) -> "KubeApiFacade":
        self._thread = threading.Thread(target=self._httpd.serve_forever,
                                        name="kube-api-facade", daemon=True)
        self._thread.start()
        return self

    def stop(self) -> None:
        self._httpd.stopping = True  # type: ignore[attr-defined]
        self._httpd.shutdown()
        self._httpd.server_close()  # release the listening socket/port
        if self._thread:
            self._thread.join(timeout=3)


def main(argv=None) -> int:
    import argparse
    parser = argparse.ArgumentParser(prog="kuberay-amd-kubeapi")
    parser.add_argument("--port", type=int, default=6443)
    parser.add_argument("--host", default="0.0.0.0")
    args = parser.parse_args(argv)
    facade = KubeApiFacade(host=args.host, port=args.port)
    print(f"kube-api facade on {facade.url}")
    facade._httpd.serve_forever()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
