"""Object store abstraction (L1 storage substrate).

Reference parity: src/object-store (OpenDAL wrapper + layers). SSTs and
manifests are keyed blobs; backends: local fs (production single-node /
cache tier) — the S3/GCS/Azure backends are the same interface with a
remote client (not reachable in this environment; the fs backend doubles as
the write-through cache the reference keeps in front of object storage,
mito2 cache/write_cache.rs). Layers wrap any backend with metrics/retry,
matching the reference's layered design.
"""

from __future__ import annotations

import os
import shutil
import time

from greptimedb_amd.utils import metrics as metrics_mod


class ObjectStore:
    def put(self, key: str, data: bytes):
        raise NotImplementedError

    def get(self, key: str) -> bytes:
        raise NotImplementedError

    def delete(self, key: str):
        raise NotImplementedError

    def list(self, prefix: str = "") -> list[str]:
        raise NotImplementedError

    def exists(self, key: str) -> bool:
        raise NotImplementedError

    def path_for(self, key: str) -> str | None:
        """Local filesystem path if this backend has one (zero-copy IO for
        parquet writers); None for remote backends."""
        return None


class FsObjectStore(ObjectStore):
    def __init__(self, root: str):
        self.root = root
        os.makedirs(root, exist_ok=True)

    def _p(self, key: str) -> str:
        p = os.path.join(self.root, key)
        os.makedirs(os.path.dirname(p), exist_ok=True)
        return p

    def put(self, key: str, data: bytes):
        tmp = self._p(key) + ".tmp"
        with open(tmp, "wb") as f:
            f.write(data)
        os.rename(tmp, self._p(key))

    def get(self, key: str) -> bytes:
        with open(self._p(key), "rb") as f:
            return f.read()

    def delete(self, key: str):
        p = self._p(key)
        if os.path.exists(p):
            os.unlink(p)

    def list(self, prefix: str = "") -> list[str]:
        out = []
        base = os.path.join(self.root, prefix)
        if not os.path.isdir(base):
            base = self.root
        for dirpath, _dirs, files in os.walk(self.root):
            for fn in files:
                key = os.path.relpath(os.path.join(dirpath, fn), self.root)
                if key.startswith(prefix):
                    out.append(key)
        return sorted(out)

    def exists(self, key: str) -> bool:
        return os.path.exists(self._p(key))

    def path_for(self, key: str) -> str:
        return self._p(key)


class MetricsLayer(ObjectStore):
    """Counts ops/bytes (reference: object-store metrics layer)."""

    def __init__(self, inner: ObjectStore):
        self.inner = inner

    def put(self, key, data):
        metrics_mod.counter("objstore_put").inc()
        metrics_mod.counter("objstore_put_bytes").inc(len(data))
        return self.inner.put(key, data)

    def get(self, key):
        metrics_mod.counter("objstore_get").inc()
        return self.inner.get(key)

    def delete(self, key):
        metrics_mod.counter("objstore_delete").inc()
        return self.inner.delete(key)

    def list(self, prefix=""):
        return self.inner.list(prefix)

    def exists(self, key):
        return self.inner.exists(key)

    def path_for(self, key):
        return self.inner.path_for(key)


class RetryLayer(ObjectStore):
    """Retries transient failures (reference: retry layer)."""

    def __init__(self, inner: ObjectStore, attempts: int = 3, backoff_s: float = 0.05):
        self.inner = inner
        self.attempts = attempts
        self.backoff_s = backoff_s

    def _retry(self, fn, *a):
        last = None
        for i in range(self.attempts):
            try:
                return fn(*a)
            except OSError as e:  # pragma: no cover - exercised via tests
                last = e
                time.sleep(self.backoff_s * (2 ** i))
        raise last

    def put(self, key, data):
        return self._retry(self.inner.put, key, data)

    def get(self, key):
        return self._retry(self.inner.get, key)

    def delete(self, key):
        return self._retry(self.inner.delete, key)

    def list(self, prefix=""):
        return self._retry(self.inner.list, prefix)

    def exists(self, key):
        return self.inner.exists(key)

    def path_for(self, key):
        return self.inner.path_for(key)
