"""Region migration: move a region between GPUs / engines.

Reference parity: src/meta-srv/src/procedure/region_migration/ — the full
state ladder open_candidate → (catchup) → downgrade_leader (write fence) →
final catchup (src/mito2/src/worker/handle_catchup.rs:35) →
upgrade_candidate → update_metadata (route flip) → close_downgraded, each
step persisted through the procedure framework for crash-resume.

Two MI355X shapes:

* `migrate_region` — same engine process, different GPU: region columns are
  device tensors, so migration is flush + tensor.to(target) over xGMI.
* `CrossEngineMigration` — between engine processes (ranks): candidate
  opens from shared storage (SST + manifest + series-log copy), catches up
  the WAL tail from the source's log, the source is fenced for the final
  delta, then routing flips (engine.route_overrides + routing_epoch bump,
  which every Ingestor observes).
"""

from __future__ import annotations

import os
import shutil

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
    """Same-process, cross-device migration (xGMI tensor move)."""

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


class CrossEngineMigration(Procedure):
    """Cross-engine (cross-rank) migration with WAL catchup + write fence.

    State: {table, region_idx, source, target, target_rank, phase}.
    `source`/`target` name engines in ENGINES (bound by the process before
    submit/recover — the reference binds datanode clients the same way at
    metasrv startup)."""

    TYPE = "cross_engine_migration"
    ENGINES: dict = {}   # name -> MitoEngine, bound by the hosting process

    def lock_key(self) -> str:
        return "region_migration"

    # ------------------------------------------------------------- helpers
    def _engines(self, state):
        src = self.ENGINES.get(state["source"])
        dst = self.ENGINES.get(state["target"])
        if src is None or dst is None:
            raise RuntimeError("engines not bound in CrossEngineMigration.ENGINES")
        return src, dst

    def _replica(self, state, src, dst):
        from greptimedb_amd.meta.replication import FollowerReplica
        return FollowerReplica(src.config.data_dir, dst,
                               tables=[state["table"]])

    # ------------------------------------------------------------- ladder
    def step(self, state: dict):
        phase = state.get("phase", "open_candidate")
        src, dst = self._engines(state)
        table, ridx = state["table"], state["region_idx"]

        if phase == "open_candidate":
            # candidate opens from shared storage: copy the region's SSTs,
            # manifest, series log and index sidecars, then open a fresh
            # Region over them (reference: open_candidate_region from the
            # shared object store)
            rep = self._replica(state, src, dst)
            rep.sync_catalog()
            s_region = src.table(table).regions[ridx]
            d_st = dst.table(table)
            d_region = d_st.regions[ridx]
            with s_region._flush_lock, s_region.lock:
                if os.path.isdir(d_region.dir):
                    d_region._series_log.close()
                    shutil.rmtree(d_region.dir)
                shutil.copytree(s_region.dir, d_region.dir)
            from greptimedb_amd.engine.region import Region
            d_st.regions[ridx] = Region(
                d_region.region_id, d_st.schema, d_region.dir,
                device=dst.config.device, append_mode=d_st.append_mode)
            state["phase"] = "catchup"
            return Status.EXECUTING, state

        if phase == "catchup":
            self._replica(state, src, dst).catchup()
            state["phase"] = "downgrade_leader"
            return Status.EXECUTING, state

        if phase == "downgrade_leader":
            # write fence on the source (reference DowngradingLeader role);
            # idempotent on resume
            src.table(table).regions[ridx].writable = False
            state["phase"] = "final_catchup"
            return Status.EXECUTING, state

        if phase == "final_catchup":
            # no writer can race this replay — the fence is up
            self._replica(state, src, dst).catchup()
            state["phase"] = "upgrade_candidate"
            return Status.EXECUTING, state

        if phase == "upgrade_candidate":
            dst.table(table).regions[ridx].writable = True
            state["phase"] = "update_metadata"
            return Status.EXECUTING, state

        if phase == "update_metadata":
            # route flip: every engine's ingestors re-resolve on the next
            # batch (routing_epoch observation in Ingestor.ingest_lines)
            tgt_rank = state.get("target_rank")
            for eng in (src, dst):
                if tgt_rank is not None:
                    overrides = getattr(eng, "route_overrides", None)
                    if overrides is None:
                        overrides = eng.route_overrides = {}
                    overrides[(table, ridx)] = tgt_rank
                eng.routing_epoch = getattr(eng, "routing_epoch", 0) + 1
                eng._save_catalog()
            from greptimedb_amd.utils.events import recorder_of
            if dst.config.record_events:
                recorder_of(dst).record("region_migration", {
                    "table": table, "region_idx": ridx,
                    "source": state["source"], "target": state["target"]})
            state["phase"] = "close_downgraded"
            return Status.EXECUTING, state

        if phase == "close_downgraded":
            region = src.table(table).regions[ridx]
            with region.lock:
                region.sst_cache.clear()
                region.flushing.clear()
                from greptimedb_amd.engine.memtable import Memtable
                region.memtable = Memtable(len(region.field_names),
                                           device=region.device)
            return Status.DONE, state

        raise RuntimeError(f"unknown phase {phase}")
