"""History-server storage backends
(reference: historyserver/pkg/storage/interface.go:10-17 + backends).

``StorageReader``/``StorageWriter`` with a filesystem backend (the
``local`` backend of the reference) and an in-memory backend for tests.
Cloud backends (S3/GCS/Azure/OSS) keep the same interface; their SDKs are
not available in this offline image, so they raise with a clear message
until configured (documented in docs/history-server.md)."""
from __future__ import annotations

import gzip
import io
import os
import threading
from typing import Dict, List


class StorageWriter:
    def write(self, path: str, data: bytes) -> None:
        raise NotImplementedError

    def append(self, path: str, data: bytes) -> None:
        raise NotImplementedError


class StorageReader:
    def read(self, path: str) -> bytes:
        raise NotImplementedError

    def list(self, prefix: str) -> List[str]:
        raise NotImplementedError

    def exists(self, path: str) -> bool:
        raise NotImplementedError


class LocalStorage(StorageReader, StorageWriter):
    """Filesystem backend (reference 'local')."""

    def __init__(self, root: str):
        self.root = root
        os.makedirs(root, exist_ok=True)
        self._lock = threading.Lock()

    def _full(self, path: str) -> str:
        full = os.path.normpath(os.path.join(self.root, path.lstrip("/")))
        if not full.startswith(os.path.abspath(self.root)):
            raise ValueError(f"path escapes storage root: {path}")
        return full

    def write(self, path: str, data: bytes) -> None:
        full = self._full(path)
        os.makedirs(os.path.dirname(full), exist_ok=True)
        tmp = full + ".tmp"
        with open(tmp, "wb") as f:
            f.write(data)
        os.replace(tmp, full)

    def append(self, path: str, data: bytes) -> None:
        full = self._full(path)
        os.makedirs(os.path.dirname(full), exist_ok=True)
        with self._lock, open(full, "ab") as f:
            f.write(data)

    def read(self, path: str) -> bytes:
        with open(self._full(path), "rb") as f:
            return f.read()

    def list(self, prefix: str) -> List[str]:
        base = self._full(prefix)
        out = []
        if not os.path.isdir(base):
            return out
        for dirpath, _, files in os.walk(base):
            for fn in files:
                full = os.path.join(dirpath, fn)
                out.append(os.path.relpath(full, self.root))
        return sorted(out)

    def exists(self, path: str) -> bool:
        return os.path.exists(self._full(path))


class MemoryStorage(StorageReader, StorageWriter):
    def __init__(self):
        self._data: Dict[str, bytes] = {}
        self._lock = threading.Lock()

    def write(self, path, data):
        with self._lock:
            self._data[path.lstrip("/")] = bytes(data)

    def append(self, path, data):
        with self._lock:
            key = path.lstrip("/")
            self._data[key] = self._data.get(key, b"") + bytes(data)

    def read(self, path):
        key = path.lstrip("/")
        with self._lock:
            if key not in self._data:
                raise FileNotFoundError(path)
            return self._data[key]

    def list(self, prefix):
        p = prefix.lstrip("/")
        with self._lock:
            return sorted(k for k in self._data if k.startswith(p))

    def exists(self, path):
        with self._lock:
            return path.lstrip("/") in self._data


class _UnavailableCloudStorage(StorageReader, StorageWriter):
    def __init__(self, backend: str):
        self.backend = backend

    def _fail(self):
        raise RuntimeError(
            f"storage backend '{self.backend}' requires its cloud SDK, which "
            "is not installed in this image; use 'local' or install the SDK "
            "(see docs/history-server.md)")

    write = append = read = list = exists = lambda self, *a, **k: self._fail()


def storage_for(backend: str, **kwargs) -> StorageReader:
    """Reference: STORAGE_BACKEND env selects s3/gcs/azure/oss/local."""
    backend = (backend or "local").lower()
    if backend == "local":
        return LocalStorage(kwargs.get("root", "/var/lib/kuberay-history"))
    if backend == "memory":
        return MemoryStorage()
    if backend in ("s3", "gcs", "azure", "oss"):
        return _UnavailableCloudStorage(backend)
    raise ValueError(f"unknown storage backend '{backend}'")


def compress(data: bytes) -> bytes:
    """reference: historyserver/pkg/compression/compression.go (gzip)."""
    buf = io.BytesIO()
    with gzip.GzipFile(fileobj=buf, mode="wb") as f:
        f.write(data)
    return buf.getvalue()


def decompress(data: bytes) -> bytes:
    return gzip.decompress(data)
