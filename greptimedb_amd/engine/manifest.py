"""Region manifest: versioned delta actions + checkpoints.

Reference parity: src/mito2/src/manifest/ — RegionMetaAction deltas
(action.rs:38-50) + checkpoint every `checkpoint_distance` versions
(manager.rs:52). We persist JSON files `%09d.json` per delta and
`checkpoint.json` snapshots; region open = last checkpoint + later deltas
(+ WAL replay above flushed_seq, done by the engine).
"""

from __future__ import annotations

import json
import os

CHECKPOINT_DISTANCE = 16


class Manifest:
    def __init__(self, dir: str):
        self.dir = dir
        os.makedirs(dir, exist_ok=True)
        self.version = 0
        self.state = {"files": {}, "flushed_seq": 0, "metadata": None, "truncated_ts": None}
        self._load()

    def _ckpt_path(self):
        return os.path.join(self.dir, "checkpoint.json")

    def _delta_path(self, v):
        return os.path.join(self.dir, f"{v:09d}.json")

    def _load(self):
        if os.path.exists(self._ckpt_path()):
            with open(self._ckpt_path()) as f:
                d = json.load(f)
            self.version = d["version"]
            self.state = d["state"]
        # apply later deltas
        deltas = sorted(
            int(f.split(".")[0]) for f in os.listdir(self.dir)
            if f.endswith(".json") and f != "checkpoint.json"
        )
        for v in deltas:
            if v <= self.version:
                continue
            with open(self._delta_path(v)) as f:
                self._apply(json.load(f))
            self.version = v

    def _apply(self, action: dict):
        kind = action["kind"]
        if kind == "change":
            self.state["metadata"] = action["metadata"]
        elif kind == "edit":
            for fmeta in action.get("files_to_add", []):
                self.state["files"][fmeta["file_id"]] = fmeta
            for fid in action.get("files_to_remove", []):
                self.state["files"].pop(fid, None)
            if action.get("flushed_seq") is not None:
                self.state["flushed_seq"] = max(self.state["flushed_seq"], action["flushed_seq"])
        elif kind == "truncate":
            self.state["files"] = {}
            self.state["flushed_seq"] = action.get("flushed_seq", self.state["flushed_seq"])

    def commit(self, action: dict):
        """Apply + persist a delta action; checkpoint periodically."""
        v = self.version + 1
        tmp = self._delta_path(v) + ".tmp"
        with open(tmp, "w") as f:
            json.dump(action, f)
            f.flush()
            os.fsync(f.fileno())
        os.rename(tmp, self._delta_path(v))
        self._apply(action)
        self.version = v
        if v % CHECKPOINT_DISTANCE == 0:
            self._checkpoint()

    def _checkpoint(self):
        tmp = self._ckpt_path() + ".tmp"
        with open(tmp, "w") as f:
            json.dump({"version": self.version, "state": self.state}, f)
            f.flush()
            os.fsync(f.fileno())
        os.rename(tmp, self._ckpt_path())
        # drop deltas folded into the checkpoint
        for fn in os.listdir(self.dir):
            if fn.endswith(".json") and fn != "checkpoint.json":
                if int(fn.split(".")[0]) <= self.version:
                    os.unlink(os.path.join(self.dir, fn))

    @property
    def files(self) -> dict:
        return self.state["files"]

    @property
    def flushed_seq(self) -> int:
        return self.state["flushed_seq"]
