"""Persistent procedure framework.

Reference parity: src/common/procedure (RFC 2023-01-03): a Procedure is a
multi-step state machine; each step's state persists to a ProcedureStore so
a crashed process resumes where it left off; lock keys serialize conflicting
procedures. Steps return Status: "executing" (persist + continue), "done",
or raise (retry / poison).
"""

from __future__ import annotations

import json
import os
import threading
import uuid


class Status:
    EXECUTING = "executing"
    DONE = "done"


class Procedure:
    """Subclass with TYPE, lock_key(), and step(state) -> (status, state)."""

    TYPE = "procedure"

    def lock_key(self) -> str:
        return ""

    def initial_state(self) -> dict:
        return {}

    def step(self, state: dict) -> tuple[str, dict]:
        raise NotImplementedError


class ProcedureStore:
    """Filesystem-backed procedure state (reference: store/ state files)."""

    def __init__(self, dir: str):
        self.dir = dir
        os.makedirs(dir, exist_ok=True)

    def save(self, pid: str, ptype: str, state: dict, status: str):
        tmp = os.path.join(self.dir, f"{pid}.tmp")
        with open(tmp, "w") as f:
            json.dump({"pid": pid, "type": ptype, "state": state,
                       "status": status}, f)
            f.flush()
            os.fsync(f.fileno())
        os.rename(tmp, os.path.join(self.dir, f"{pid}.json"))

    def load_all(self) -> list[dict]:
        out = []
        for fn in sorted(os.listdir(self.dir)):
            if fn.endswith(".json"):
                with open(os.path.join(self.dir, fn)) as f:
                    out.append(json.load(f))
        return out

    def remove(self, pid: str):
        p = os.path.join(self.dir, f"{pid}.json")
        if os.path.exists(p):
            os.unlink(p)


class ProcedureManager:
    def __init__(self, store_dir: str):
        self.store = ProcedureStore(store_dir)
        self.registry: dict[str, type[Procedure]] = {}
        self._locks: dict[str, threading.Lock] = {}
        self._guard = threading.Lock()

    def register(self, cls: type[Procedure]):
        self.registry[cls.TYPE] = cls
        return cls

    def _lock_for(self, key: str) -> threading.Lock:
        with self._guard:
            return self._locks.setdefault(key, threading.Lock())

    def submit(self, proc: Procedure, state: dict | None = None,
               pid: str | None = None) -> str:
        """Run a procedure to completion (synchronously), persisting each
        step. Returns procedure id."""
        pid = pid or uuid.uuid4().hex
        state = dict(state if state is not None else proc.initial_state())
        lock = self._lock_for(proc.lock_key() or pid)
        with lock:
            self.store.save(pid, proc.TYPE, state, Status.EXECUTING)
            while True:
                status, state = proc.step(state)
                self.store.save(pid, proc.TYPE, state, status)
                if status == Status.DONE:
                    self.store.remove(pid)
                    return pid

    def recover(self) -> list[str]:
        """Resume procedures persisted as executing (crash recovery)."""
        resumed = []
        for rec in self.store.load_all():
            cls = self.registry.get(rec["type"])
            if cls is None or rec["status"] == Status.DONE:
                self.store.remove(rec["pid"])
                continue
            proc = cls()
            self.submit(proc, state=rec["state"], pid=rec["pid"])
            resumed.append(rec["pid"])
        return resumed
