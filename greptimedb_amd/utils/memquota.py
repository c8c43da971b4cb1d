"""Scan memory quota (ref: src/common/memory-manager — semaphore-based
memory quota used by the reference's scan memory pool, lib.rs:17).

Large raw scans materialize host-side result columns; concurrent serving
queries could otherwise stack unbounded host allocations. A quota is a
counting semaphore over bytes: queries acquire their estimated result
footprint and block (bounded) until capacity frees; a single request
larger than the whole quota fails loudly instead of OOM-ing the process.
"""

from __future__ import annotations

import threading

from greptimedb_amd.utils.errors import GreptimeError


class ResourceExhausted(GreptimeError):
    pass


class MemoryQuota:
    def __init__(self, total_bytes: int):
        self.total = int(total_bytes)
        self._free = self.total
        self._cv = threading.Condition()

    def acquire(self, nbytes: int, timeout_s: float = 30.0):
        """Reserve bytes; returns a release handle (context manager)."""
        nbytes = int(nbytes)
        if nbytes > self.total:
            raise ResourceExhausted(
                f"scan needs {nbytes >> 20} MiB, quota is "
                f"{self.total >> 20} MiB — narrow the query")
        with self._cv:
            ok = self._cv.wait_for(lambda: self._free >= nbytes,
                                   timeout=timeout_s)
            if not ok:
                raise ResourceExhausted(
                    f"scan memory quota exhausted waiting for "
                    f"{nbytes >> 20} MiB")
            self._free -= nbytes
        return _Reservation(self, nbytes)

    def _release(self, nbytes: int):
        with self._cv:
            self._free += nbytes
            self._cv.notify_all()

    @property
    def free(self) -> int:
        with self._cv:
            return self._free


class _Reservation:
    def __init__(self, quota: MemoryQuota, nbytes: int):
        self._quota = quota
        self._n = nbytes

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.release()

    def release(self):
        if self._quota is not None:
            self._quota._release(self._n)
            self._quota = None
