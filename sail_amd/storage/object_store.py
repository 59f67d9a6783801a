"""Object-store registry: URI scheme+authority -> store instance.

The reference keys stores by (scheme, authority, session fingerprint) and
supports s3/gcs/azure/http/hdfs/huggingface/memory/local
(ref: crates/sail-object-store/src/registry.rs:25-40). Here:

  * local   — bare paths and file:// (direct OS filesystem)
  * memory  — memory:// in-process store (tests, temp artifacts)
  * s3/gs/abfs/http(s)/hf — fsspec-backed when the protocol's driver is
    importable; the image has no network, so these construct lazily and
    surface the driver error on first IO, like the reference's lazy-init
    layer (src/layers/).

Stores expose a small byte-level API plus `as_local(path)` which stages a
remote object to a local file so the pyarrow/GPU readers can run on it.
"""
from __future__ import annotations

import io
import os
import threading
from typing import Dict, List, Optional, Tuple
from urllib.parse import urlparse


def split_uri(uri: str) -> Tuple[str, str, str]:
    """(scheme, authority, path); bare paths get scheme 'file'."""
    if "://" not in uri:
        return "file", "", uri
    p = urlparse(uri)
    return p.scheme.lower(), p.netloc, p.path


class ObjectStore:
    scheme = ""

    def open_read(self, path: str):
        raise NotImplementedError

    def write_bytes(self, path: str, data: bytes):
        raise NotImplementedError

    def read_bytes(self, path: str) -> bytes:
        with self.open_read(path) as f:
            return f.read()

    def exists(self, path: str) -> bool:
        raise NotImplementedError

    def list(self, prefix: str) -> List[str]:
        raise NotImplementedError

    def delete(self, path: str):
        raise NotImplementedError

    def as_local(self, path: str) -> str:
        """Local filesystem path holding this object (staged if remote)."""
        import hashlib
        import tempfile

        key = hashlib.sha1(path.encode()).hexdigest()[:16]
        target = os.path.join(tempfile.gettempdir(), "sail_store_cache",
                              f"{key}-{os.path.basename(path)}")
        os.makedirs(os.path.dirname(target), exist_ok=True)
        # always restage: the object may have been overwritten since the
        # last read (a path-keyed cache would serve stale bytes)
        with open(target, "wb") as f:
            f.write(self.read_bytes(path))
        return target


class LocalStore(ObjectStore):
    scheme = "file"

    def open_read(self, path: str):
        return open(path, "rb")

    def write_bytes(self, path: str, data: bytes):
        d = os.path.dirname(path)
        if d:
            os.makedirs(d, exist_ok=True)
        with open(path, "wb") as f:
            f.write(data)

    def exists(self, path: str) -> bool:
        return os.path.exists(path)

    def list(self, prefix: str) -> List[str]:
        out = []
        root = prefix if os.path.isdir(prefix) else os.path.dirname(prefix)
        for base, _dirs, files in os.walk(root):
            for fn in files:
                p = os.path.join(base, fn)
                if p.startswith(prefix):
                    out.append(p)
        return sorted(out)

    def delete(self, path: str):
        os.remove(path)

    def as_local(self, path: str) -> str:
        return path


class MemoryStore(ObjectStore):
    scheme = "memory"

    def __init__(self):
        self._objs: Dict[str, bytes] = {}
        self._lock = threading.Lock()

    def open_read(self, path: str):
        with self._lock:
            if path not in self._objs:
                raise FileNotFoundError(f"memory://{path}")
            return io.BytesIO(self._objs[path])

    def write_bytes(self, path: str, data: bytes):
        with self._lock:
            self._objs[path] = bytes(data)

    def exists(self, path: str) -> bool:
        with self._lock:
            return path in self._objs

    def list(self, prefix: str) -> List[str]:
        with self._lock:
            return sorted(k for k in self._objs if k.startswith(prefix))

    def delete(self, path: str):
        with self._lock:
            self._objs.pop(path, None)


class FsspecStore(ObjectStore):
    """Adapter over an fsspec filesystem (s3/gs/abfs/http/hf...). The
    filesystem is constructed lazily: registration never fails, the first
    IO surfaces the missing-driver/network error."""

    def __init__(self, scheme: str, authority: str,
                 options: Optional[Dict[str, str]] = None):
        self.scheme = scheme
        self.authority = authority
        self.options = options or {}
        self._fs = None

    def _filesystem(self):
        if self._fs is None:
            import fsspec

            self._fs = fsspec.filesystem(self.scheme, **self.options)
        return self._fs

    def _full(self, path: str) -> str:
        return f"{self.authority}{path}" if self.authority else path.lstrip("/")

    def open_read(self, path: str):
        return self._filesystem().open(self._full(path), "rb")

    def write_bytes(self, path: str, data: bytes):
        with self._filesystem().open(self._full(path), "wb") as f:
            f.write(data)

    def exists(self, path: str) -> bool:
        return self._filesystem().exists(self._full(path))

    def list(self, prefix: str) -> List[str]:
        return sorted(self._filesystem().find(self._full(prefix)))

    def delete(self, path: str):
        self._filesystem().rm(self._full(path))


class ObjectStoreRegistry:
    """Stores keyed by (scheme, authority) — one instance per key per
    registry, mirroring the reference's session-fingerprinted cache."""

    FSSPEC_SCHEMES = ("s3", "s3a", "gs", "gcs", "abfs", "abfss", "az",
                      "http", "https", "hf", "hdfs")

    def __init__(self):
        self._stores: Dict[Tuple[str, str], ObjectStore] = {}
        self._lock = threading.Lock()
        self._local = LocalStore()

    def register(self, scheme: str, authority: str, store: ObjectStore):
        with self._lock:
            self._stores[(scheme.lower(), authority)] = store

    def for_uri(self, uri: str) -> Tuple[ObjectStore, str]:
        """(store, path-within-store) for a URI or bare path."""
        scheme, authority, path = split_uri(uri)
        if scheme in ("", "file"):
            return self._local, path
        key = (scheme, authority)
        with self._lock:
            st = self._stores.get(key)
            if st is None:
                if scheme == "memory":
                    st = MemoryStore()
                elif scheme in self.FSSPEC_SCHEMES:
                    st = FsspecStore(scheme, authority)
                else:
                    raise ValueError(f"no object store for scheme {scheme}://")
                self._stores[key] = st
        return st, path


_GLOBAL = ObjectStoreRegistry()


def global_registry() -> ObjectStoreRegistry:
    return _GLOBAL


def resolve_local(uri: str) -> str:
    """URI -> local filesystem path (staging remote objects); passthrough
    for plain paths. The datasource readers call this so any registered
    scheme can feed the parquet/CSV/GPU decode paths."""
    store, path = _GLOBAL.for_uri(uri)
    return store.as_local(path)


def expand_to_local(uri: str) -> List[str]:
    """Expand a remote URI (object or prefix) to staged local file paths."""
    store, path = _GLOBAL.for_uri(uri)
    if store.exists(path) and not path.endswith("/"):
        objs = [path]
        listed = store.list(path)
        if listed and listed != [path]:
            objs = listed
    else:
        objs = store.list(path)
    if not objs:
        raise FileNotFoundError(uri)
    return [store.as_local(o) for o in objs]
