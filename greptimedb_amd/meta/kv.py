"""Metadata KV with CAS, TTL leases and leader election.

Reference parity: the reference's metasrv keeps cluster metadata in an
etcd-style KV (src/common/meta kv_backend: range/put/CAS/lease, election
in src/meta-srv/src/election). There is no external etcd in this
environment, so this is a crash-safe single-node implementation over a
lock-protected JSON file — the same API shape (get/put/cas/delete/range,
lease grant/keepalive/expire, campaign/resign) so a networked backend can
slot in behind it. Multi-process safe via fcntl file locking (the 8
one-process-per-GPU ranks of a node share the filesystem).
"""

from __future__ import annotations

import fcntl
import json
import os
import time


class MetaKV:
    def __init__(self, path: str):
        self.path = path
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        if not os.path.exists(path):
            self._write({"kv": {}, "leases": {}, "rev": 0})

    # ------------------------------------------------------------ file ops

    def _locked(self):
        f = open(self.path, "r+")
        fcntl.flock(f, fcntl.LOCK_EX)
        return f

    def _read_f(self, f) -> dict:
        f.seek(0)
        return json.load(f)

    def _write_f(self, f, state: dict):
        state["rev"] += 1
        tmp = self.path + ".tmp"
        with open(tmp, "w") as t:
            json.dump(state, t)
        os.replace(tmp, self.path)

    def _write(self, state: dict):
        tmp = self.path + ".tmp"
        with open(tmp, "w") as t:
            json.dump(state, t)
        os.replace(tmp, self.path)

    def _expire(self, state: dict):
        now = time.time()
        dead = [lid for lid, l in state["leases"].items()
                if l["deadline"] < now]
        for lid in dead:
            for k in state["leases"][lid]["keys"]:
                state["kv"].pop(k, None)
            del state["leases"][lid]
        return bool(dead)

    # ------------------------------------------------------------------ kv

    def get(self, key: str):
        with self._locked() as f:
            state = self._read_f(f)
            if self._expire(state):
                self._write_f(f, state)
            v = state["kv"].get(key)
            return v["value"] if v else None

    def put(self, key: str, value, lease: str | None = None):
        with self._locked() as f:
            state = self._read_f(f)
            self._expire(state)
            state["kv"][key] = {"value": value, "lease": lease}
            if lease is not None:
                if lease not in state["leases"]:
                    raise KeyError(f"unknown lease {lease}")
                keys = state["leases"][lease]["keys"]
                if key not in keys:
                    keys.append(key)
            self._write_f(f, state)

    def cas(self, key: str, expect, value, lease: str | None = None) -> bool:
        """Compare-and-swap: expect None = key must be absent."""
        with self._locked() as f:
            state = self._read_f(f)
            self._expire(state)
            cur = state["kv"].get(key)
            cur_v = cur["value"] if cur else None
            if cur_v != expect:
                self._write_f(f, state)
                return False
            state["kv"][key] = {"value": value, "lease": lease}
            if lease is not None:
                if lease not in state["leases"]:
                    raise KeyError(f"unknown lease {lease}")
                keys = state["leases"][lease]["keys"]
                if key not in keys:
                    keys.append(key)
            self._write_f(f, state)
            return True

    def delete(self, key: str) -> bool:
        with self._locked() as f:
            state = self._read_f(f)
            self._expire(state)
            had = state["kv"].pop(key, None) is not None
            self._write_f(f, state)
            return had

    def range(self, prefix: str) -> dict:
        with self._locked() as f:
            state = self._read_f(f)
            if self._expire(state):
                self._write_f(f, state)
            return {k: v["value"] for k, v in state["kv"].items()
                    if k.startswith(prefix)}

    # -------------------------------------------------------------- leases

    def grant_lease(self, ttl_s: float, lease_id: str | None = None) -> str:
        with self._locked() as f:
            state = self._read_f(f)
            self._expire(state)
            lid = lease_id or f"lease-{state['rev']}-{os.getpid()}"
            state["leases"][lid] = {"ttl": ttl_s,
                                    "deadline": time.time() + ttl_s,
                                    "keys": []}
            self._write_f(f, state)
            return lid

    def keepalive(self, lease_id: str) -> bool:
        with self._locked() as f:
            state = self._read_f(f)
            self._expire(state)
            l = state["leases"].get(lease_id)
            if l is None:
                self._write_f(f, state)
                return False
            l["deadline"] = time.time() + l["ttl"]
            self._write_f(f, state)
            return True

    def revoke(self, lease_id: str):
        with self._locked() as f:
            state = self._read_f(f)
            l = state["leases"].pop(lease_id, None)
            if l:
                for k in l["keys"]:
                    state["kv"].pop(k, None)
            self._write_f(f, state)


class Election:
    """Leader election on the KV (ref meta-srv election: leader key with a
    lease; the holder is the leader, expiry frees the seat)."""

    def __init__(self, kv: MetaKV, key: str, node: str, ttl_s: float = 5.0):
        self.kv = kv
        self.key = key
        self.node = node
        self.ttl_s = ttl_s
        self.lease: str | None = None

    def campaign(self) -> bool:
        """Try to take (or confirm) leadership; non-blocking."""
        cur = self.kv.get(self.key)
        if cur == self.node:
            return self.kv.keepalive(self.lease) if self.lease else True
        if cur is not None:
            return False
        self.lease = self.kv.grant_lease(self.ttl_s)
        if self.kv.cas(self.key, None, self.node, lease=self.lease):
            return True
        self.kv.revoke(self.lease)
        self.lease = None
        return False

    def leader(self) -> str | None:
        return self.kv.get(self.key)

    def resign(self):
        if self.lease:
            self.kv.revoke(self.lease)
            self.lease = None
