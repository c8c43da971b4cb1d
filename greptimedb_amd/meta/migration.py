"""Region migration: move a region between GPUs.

Reference parity: src/meta-srv/src/procedure/region_migration/ — the
state machine open_candidate → flush_leader → downgrade_leader →
upgrade_candidate → update_metadata → close_downgraded. MI355X mapping: a
region's columns are device tensors, so "migration" is flush (bound the
mutable state) + tensor.to(target) over xGMI + catalog update. Each step
persists through the procedure framework for crash-resume.
"""

from __future__ import annotations

import torch

from greptimedb_amd.meta.procedure import Procedure, Status


def _move_batch(batch, device: str):
    batch.ts = batch.ts.to(device)
    batch.series = batch.series.to(device)
    batch.fields = batch.fields.to(device)
    if batch.seq is not None:
        batch.seq = batch.seq.to(device)
    for seg in getattr(batch, "text_index", {}).values():
        seg.rows = seg.rows.to(device)


def migrate_region(engine, table_name: str, region_idx: int, target_device: str):
    """Synchronous migration of one region's resident data to another
    device. Writes are expected to be quiesced by the caller (reference:
    downgrade_leader pauses the leader before catchup)."""
    st = engine.table(table_name)
    region = st.regions[region_idx]
    region.flush()  # flush_leader: bound mutable state (memtable → SST)
    with region.lock:
        for batch in region.sst_cache.values():
            _move_batch(batch, target_device)
        mem = region.memtable
        mem.ts = mem.ts.to(target_device)
        mem.series = mem.series.to(target_device)
        mem.fields = mem.fields.to(target_device)
        mem.device = target_device
        region.device = target_device
    engine._save_catalog()
    return region


class RegionMigrationProcedure(Procedure):
    TYPE = "region_migration"

    def __init__(self, engine=None):
        self.engine = engine

    def lock_key(self) -> str:
        return "region_migration"

    def step(self, state: dict):
        phase = state.get("phase", "flush_leader")
        if self.engine is None:
            raise RuntimeError("engine not bound")
        if phase == "flush_leader":
            st = self.engine.table(state["table"])
            st.regions[state["region_idx"]].flush()
            state["phase"] = "move"
            return Status.EXECUTING, state
        if phase == "move":
            migrate_region(self.engine, state["table"], state["region_idx"],
                           state["target"])
            state["phase"] = "update_metadata"
            return Status.EXECUTING, state
        if phase == "update_metadata":
            self.engine._save_catalog()
            return Status.DONE, state
        raise RuntimeError(f"unknown phase {phase}")
