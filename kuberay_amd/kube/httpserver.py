"""Kubernetes REST façade over the in-memory API server.

Speaks enough of the kube-apiserver HTTP surface (typed paths, status
subresource, labelSelector, chunked ``?watch=true`` streams) for the
deployable REST stack (kuberay_amd/kube/rest.py RestClient +
RestApiServerAdapter) to run against it unchanged — the integration-test
analog of envtest's real apiserver, and a usable standalone control-plane
endpoint (``python -m kuberay_amd.kube.httpserver``).
"""
from __future__ import annotations

import base64
import binascii
import json
import re
import threading
import time
import uuid
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Any, Dict, List, Optional, Tuple
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


# dynamic kinds (CRDs the static map doesn't know: PodGroup, cert-manager
# objects, ...) learned from POST bodies / store contents, keyed like the
# static map
_DYNAMIC_KINDS: Dict[Tuple[str, str], str] = {}


def _plural_of(kind: str) -> str:
    # mirror of RestClient._path's convention for unknown kinds
    return kind.lower() + ("es" if kind.lower().endswith("s") else "s")


def _parse_path(path: str, store=None, body_kind: Optional[str] = None,
                ) -> Optional[Tuple[str, Optional[str], Optional[str], Optional[str]]]:
    m = _PATH_RE.match(path)
    if not m:
        return None
    key = (m.group("prefix"), m.group("plural"))
    kind = _KIND_BY_PREFIX_PLURAL.get(key) or _DYNAMIC_KINDS.get(key)
    if kind is None and store is not None:
        for k in store.kinds():
            if _plural_of(k) == key[1]:
                kind = _DYNAMIC_KINDS[key] = k
                break
    if kind is None and body_kind and _plural_of(body_kind) == key[1]:
        kind = _DYNAMIC_KINDS[key] = body_kind
    if kind is None:
        return None
    name, sub = m.group("name"), m.group("sub")
    if name == "status":  # /plural/name/status matched oddly
        name, sub = None, "status"
    return kind, m.group("ns"), name, sub


class JsonPatchTestFailed(Exception):
    """RFC 6902 `test` op failed → 409 like the real apiserver."""


def _resolve_pointer(doc: Any, path: str, *, parent: bool = False):
    """JSON-pointer walk; returns (container, last_token)."""
    if path == "":
        return None, None
    tokens = [t.replace("~1", "/").replace("~0", "~")
              for t in path.lstrip("/").split("/")]
    node = doc
    walk = tokens[:-1] if parent else tokens
    for token in walk:
        if isinstance(node, list):
            node = node[int(token)]
        elif isinstance(node, dict):
            if token not in node:
                raise ValueError(f"path {path} not found")
            node = node[token]
        else:
            raise ValueError(f"path {path} not traversable")
    return node, tokens[-1]


def apply_json_patch(doc: Dict[str, Any],
                     ops: List[Dict[str, Any]]) -> Dict[str, Any]:
    """RFC 6902 add/remove/replace/test/copy/move on a deep copy."""
    out = json.loads(json.dumps(doc))
    if not isinstance(ops, list):
        raise ValueError("json-patch body must be an array of operations")
    for op in ops:
        kind, path = op.get("op"), op.get("path", "")
        container, last = _resolve_pointer(out, path, parent=True)
        if kind == "add":
            if isinstance(container, list):
                idx = len(container) if last == "-" else int(last)
                container.insert(idx, op.get("value"))
            else:
                container[last] = op.get("value")
        elif kind == "replace":
            if isinstance(container, list):
                container[int(last)] = op.get("value")
            else:
                if last not in container:
                    raise ValueError(f"replace path {path} not found")
                container[last] = op.get("value")
        elif kind == "remove":
            if isinstance(container, list):
                container.pop(int(last))
            else:
                if last not in container:
                    raise ValueError(f"remove path {path} not found")
                del container[last]
        elif kind == "test":
            node, _ = _resolve_pointer(out, path)
            if node != op.get("value"):
                raise JsonPatchTestFailed(
                    f"test failed at {path}: {node!r} != {op.get('value')!r}")
        elif kind in ("copy", "move"):
            src_node, _ = _resolve_pointer(out, op.get("from", ""))
            value = json.loads(json.dumps(src_node))
            if kind == "move":
                src_parent, src_last = _resolve_pointer(
                    out, op.get("from", ""), parent=True)
                if isinstance(src_parent, list):
                    src_parent.pop(int(src_last))
                else:
                    del src_parent[src_last]
            if isinstance(container, list):
                idx = len(container) if last == "-" else int(last)
                container.insert(idx, value)
            else:
                container[last] = value
        else:
            raise ValueError(f"unsupported json-patch op {kind!r}")
    return out


# strategic-merge list merge keys (subset of the k8s patchMergeKey tags
# that matter for Ray pod specs)
_MERGE_KEYS = {"containers": "name", "initContainers": "name",
               "volumes": "name", "env": "name", "volumeMounts": "name",
               "ports": "containerPort", "imagePullSecrets": "name",
               "workerGroupSpecs": "groupName"}


def strategic_merge(current: Any, patch: Any, field: str = "") -> Any:
    """Strategic merge patch: dict-recursive like merge patch, but lists
    with a known merge key merge element-wise instead of replacing, and
    ``$patch: delete`` directives remove elements."""
    if isinstance(patch, dict) and isinstance(current, dict):
        out = dict(current)
        for key, val in patch.items():
            if val is None:
                out.pop(key, None)
            elif key in out:
                out[key] = strategic_merge(out[key], val, key)
            elif isinstance(val, dict):
                # RFC 7386 MergePatch(target, patch) with target absent:
                # recurse into {} so nested nulls delete (never materialize)
                out[key] = strategic_merge({}, val, key)
            else:
                out[key] = val
        return out
    if isinstance(patch, list) and isinstance(current, list) and \
            field in _MERGE_KEYS:
        merge_key = _MERGE_KEYS[field]
        out = list(current)
        index = {item.get(merge_key): i for i, item in enumerate(out)
                 if isinstance(item, dict)}
        for elem in patch:
            if not isinstance(elem, dict):
                out.append(elem)
                continue
            key_val = elem.get(merge_key)
            if elem.get("$patch") == "delete":
                out = [o for o in out
                       if not (isinstance(o, dict)
                               and o.get(merge_key) == key_val)]
                index = {item.get(merge_key): i for i, item in enumerate(out)
                         if isinstance(item, dict)}
                continue
            if key_val in index:
                out[index[key_val]] = strategic_merge(out[index[key_val]],
                                                      elem)
            else:
                out.append(strategic_merge({}, elem))
                index[key_val] = len(out) - 1
        return out
    if isinstance(patch, dict):
        # replacing a non-dict with a dict: same null-stripping recursion
        return strategic_merge({}, patch, field)
    return patch


class _Handler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"
    # headers and body are separate writes; without TCP_NODELAY the body
    # segment waits for the client's delayed ACK of the header segment
    # (~40 ms/op — this was the sharded scale-out bottleneck)
    disable_nagle_algorithm = True
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
        if not length:
            return {}
        try:
            return json.loads(self.rfile.read(length))
        except ValueError as e:
            raise ApiError(400, f"invalid JSON body: {e}")

    # ------------------------------------------------------------------
    def do_GET(self):  # noqa: N802
        url = urlparse(self.path)
        parsed = _parse_path(url.path, store=self.store)
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
            rv = params.get("resourceVersion", [None])[0]
            bookmarks = params.get("allowWatchBookmarks",
                                   ["false"])[0] == "true"
            return self._stream_watch(kind, rv, bookmarks)
        selector = None
        if "labelSelector" in params:
            selector = dict(kv.split("=", 1)
                            for kv in params["labelSelector"][0].split(","))
        items = self.store.list(kind, ns, selector)
        # chunked lists: ?limit=N&continue=token. Real-apiserver
        # semantics: the continue token pins a consistent SNAPSHOT (etcd
        # revision) — later pages never skip or duplicate items mutated
        # between requests, and an expired token answers 410 Expired.
        list_meta = {"resourceVersion": str(self.store.current_rv)}
        limit = int(params.get("limit", ["0"])[0] or 0)
        if limit > 0:
            snaps = getattr(self.server, "_page_snaps", None)
            if snaps is None:
                snaps = self.server._page_snaps = {}  # type: ignore[attr-defined]
            token = params.get("continue", [None])[0]
            if token:
                try:
                    snap_id, offset = base64.b64decode(
                        token).decode().split(":", 1)
                    offset = int(offset)
                except (ValueError, binascii.Error):
                    return self._send_json(400, {
                        "kind": "Status", "status": "Failure",
                        "reason": "BadRequest",
                        "message": "invalid continue token", "code": 400})
                snap = snaps.get(snap_id)
                if snap is None or time.time() - snap[2] > 300:
                    return self._send_json(410, {
                        "kind": "Status", "status": "Failure",
                        "reason": "Expired",
                        "message": "The provided continue parameter is too "
                                   "old to display a consistent list result.",
                        "code": 410})
                items, rv, _ = snap
            else:
                snap_id = uuid.uuid4().hex[:12]
                rv = str(self.store.current_rv)
                offset = 0
                if len(items) > limit:
                    # bound the snapshot table: drop expired, then oldest
                    now = time.time()
                    # concurrent handler threads mutate snaps: iterate a
                    # copy, pop defensively
                    for k, v in list(snaps.items()):
                        if now - v[2] > 300:
                            snaps.pop(k, None)
                    while len(snaps) >= 64:
                        try:
                            snaps.pop(next(iter(snaps)), None)
                        except (StopIteration, RuntimeError):
                            break
                    snaps[snap_id] = (items, rv, now)
            list_meta["resourceVersion"] = rv
            page = items[offset:offset + limit]
            if offset + limit < len(items):
                list_meta["continue"] = base64.b64encode(
                    f"{snap_id}:{offset + limit}".encode()).decode()
            else:
                snaps.pop(snap_id, None)  # fully consumed
            items = page
        self._send_json(200, {"kind": f"{kind}List", "apiVersion": "v1",
                              "metadata": list_meta, "items": items})

    def _stream_watch(self, kind: str, resource_version: Optional[str],
                      bookmarks: bool) -> None:
        """Watch with kube-apiserver semantics: resume from rv (replay from
        the event history), 410 Gone when the rv predates retained history,
        optional BOOKMARK events carrying the current rv."""
        start_rv = 0
        if resource_version:
            try:
                start_rv = int(resource_version)
            except ValueError:
                start_rv = 0
        # live watcher FIRST, replay second: the reverse order loses every
        # event that lands between the history read and the registration
        # (the rv-dedup below drops the overlap instead)
        watcher = self.store.watch({kind})
        replay = []
        if resource_version:
            replay = self.store.events_since(start_rv, {kind})
            if replay is None:
                # too old — client must re-list (Expired)
                self.store.stop_watch(watcher)
                return self._send_json(410, {
                    "kind": "Status", "apiVersion": "v1",
                    "status": "Failure", "reason": "Expired",
                    "message": f"too old resource version: {start_rv}",
                    "code": 410})
        try:
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Transfer-Encoding", "chunked")
            self.end_headers()

            def emit(event_type, obj):
                line = json.dumps({"type": event_type, "object": obj}) + "\n"
                data = line.encode()
                self.wfile.write(f"{len(data):x}\r\n".encode() + data + b"\r\n")
                self.wfile.flush()

            seen_rv = start_rv
            for event_type, obj in replay:
                emit(event_type, obj)
                try:
                    seen_rv = max(seen_rv, int(
                        obj.get("metadata", {}).get("resourceVersion") or 0))
                except ValueError:
                    pass
            idle = 0.0
            while not self.server.stopping:  # type: ignore[attr-defined]
                ev = watcher.next(timeout=0.5)
                if ev is None:
                    idle += 0.5
                    if bookmarks and idle >= 1.0:
                        idle = 0.0
                        emit("BOOKMARK", {
                            "kind": kind, "apiVersion": "v1",
                            "metadata": {"resourceVersion":
                                         str(self.store.current_rv)}})
                    continue
                idle = 0.0
                event_type, obj = ev
                # skip events already delivered via replay
                try:
                    rv = int(obj.get("metadata", {})
                             .get("resourceVersion") or 0)
                except ValueError:
                    rv = 0
                if event_type != "DELETED" and 0 < rv <= seen_rv:
                    continue
                emit(event_type, obj)
            self.wfile.write(b"0\r\n\r\n")
        except (BrokenPipeError, ConnectionResetError, OSError):
            pass
        finally:
            watcher.stop()

    def do_POST(self):  # noqa: N802
        try:
            obj = self._body()
        except ApiError as e:
            return self._send_error_status(e)
        parsed = _parse_path(urlparse(self.path).path, store=self.store,
                             body_kind=obj.get("kind"))
        if parsed is None:
            return self._send_json(404, {"message": "unknown path"})
        kind, ns, _, _ = parsed
        obj.setdefault("kind", kind)
        obj.setdefault("metadata", {}).setdefault("namespace", ns or "default")
        try:
            self._send_json(201, self.store.create(obj))
        except ApiError as e:
            self._send_error_status(e)

    def do_PUT(self):  # noqa: N802
        try:
            obj = self._body()
        except ApiError as e:
            return self._send_error_status(e)
        parsed = _parse_path(urlparse(self.path).path, store=self.store,
                             body_kind=obj.get("kind"))
        if parsed is None:
            return self._send_json(404, {"message": "unknown path"})
        kind, ns, name, sub = parsed
        obj.setdefault("kind", kind)
        obj.setdefault("metadata", {}).setdefault("namespace", ns or "default")
        if name:
            obj["metadata"]["name"] = name
        try:
            self._send_json(200, self.store.update(obj, subresource=sub))
        except ApiError as e:
            self._send_error_status(e)

    def do_PATCH(self):  # noqa: N802
        """Dispatch on Content-Type like the real apiserver:
        merge-patch (RFC 7386), json-patch (RFC 6902) and
        strategic-merge-patch (list merge by the `name` merge key)."""
        parsed = _parse_path(urlparse(self.path).path, store=self.store)
        if parsed is None:
            return self._send_json(404, {"message": "unknown path"})
        kind, ns, name, sub = parsed
        content_type = (self.headers.get("Content-Type") or "").split(";")[0]
        try:
            body = self._body()
        except ApiError as e:
            return self._send_error_status(e)
        try:
            if content_type == "application/json-patch+json":
                current = self.store.get(kind, ns or "default", name)
                patched = apply_json_patch(current, body)
                # PUT with the read rv → optimistic concurrency holds
                self._send_json(200, self.store.update(patched,
                                                       subresource=sub))
            elif content_type == "application/strategic-merge-patch+json":
                current = self.store.get(kind, ns or "default", name)
                patched = strategic_merge(current, body)
                self._send_json(200, self.store.update(patched,
                                                       subresource=sub))
            else:  # merge patch (default)
                self._send_json(200, self.store.patch_merge(
                    kind, ns or "default", name, body, subresource=sub))
        except JsonPatchTestFailed as e:
            self._send_json(409, {
                "kind": "Status", "apiVersion": "v1", "status": "Failure",
                "reason": "Conflict", "message": str(e), "code": 409})
        except ValueError as e:
            self._send_json(422, {
                "kind": "Status", "apiVersion": "v1", "status": "Failure",
                "reason": "Invalid", "message": str(e), "code": 422})
        except ApiError as e:
            self._send_error_status(e)

    def do_DELETE(self):  # noqa: N802
        parsed = _parse_path(urlparse(self.path).path, store=self.store)
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
