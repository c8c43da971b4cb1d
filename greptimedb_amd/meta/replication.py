"""Region replication: leader/follower roles + follower catchup (P7).

Reference parity: store-api region roles Leader/Follower/StagingLeader
(src/store-api/src/region_engine.rs:209-222) + mito2 follower catchup via
WAL replay (src/mito2/src/worker/handle_catchup.rs:35, RegionCatchupRequest)
and the region-lease role model.

MI355X redesign: one engine process per GPU on one node; replication is
WAL shipping over the shared filesystem. A follower engine (its regions on
a DIFFERENT GPU) tails the leader engine's WAL directory and series logs
and applies entries above its own high-water mark into its own device
memtables. Failover is then a pure role flip: the follower's copy is
already resident in its GPU's HBM — no data movement on promotion
(contrast: the reference replays Kafka WAL on catchup).
"""

from __future__ import annotations

import os
import struct

import numpy as np

from greptimedb_amd.engine.engine import MitoEngine
from greptimedb_amd.engine.wal import decode_batch
from greptimedb_amd import _native


class FollowerReplica:
    """Read-only follower of (a subset of) a leader engine's tables.

    The follower engine must hold the same table layout (create it with the
    same schema/partitions, or let `sync_catalog()` copy the leader's
    catalog). `catchup()` is idempotent and incremental — call it on a
    timer (the reference's region lease/heartbeat cadence) or before a
    read that needs freshness."""

    def __init__(self, leader_dir: str, engine: MitoEngine,
                 tables: list[str] | None = None):
        self.leader_dir = leader_dir
        self.engine = engine
        self.tables = tables
        self.applied_seq = 0
        self.role = "follower"
        self._series_offsets: dict[int, int] = {}   # region_id → bytes read

    # ------------------------------------------------------------ catalog

    def sync_catalog(self):
        """Create any leader tables missing on the follower (same schema,
        same region count — region ids must match for WAL routing)."""
        import json
        from greptimedb_amd.models.schema import TableSchema
        path = os.path.join(self.leader_dir, "catalog.json")
        if not os.path.exists(path):
            return
        with open(path) as f:
            cat = json.load(f)
        for td in cat["tables"]:
            schema = TableSchema.from_dict(td["schema"])
            if self.tables is not None and schema.name not in self.tables:
                continue
            if schema.name in self.engine.tables:
                continue
            self.engine.create_table(schema, n_regions=td["n_regions"],
                                     append_mode=td["append_mode"],
                                     if_not_exists=True)
        self.engine.next_table_id = max(self.engine.next_table_id,
                                        cat["next_table_id"])

    # ------------------------------------------------------------ catchup

    def _sync_series(self, region, leader_region_dir: str):
        """Tail the leader's series.log so follower codes stay aligned
        (both sides assign dense codes in log order)."""
        path = os.path.join(leader_region_dir, "series.log")
        if not os.path.exists(path):
            return
        off = self._series_offsets.get(region.region_id, 0)
        size = os.path.getsize(path)
        if size <= off:
            return
        with open(path, "rb") as f:
            f.seek(off)
            buf = f.read(size - off)
        pos = 0
        with region.lock:
            while pos + 4 <= len(buf):
                (ln,) = struct.unpack_from("<I", buf, pos)
                if pos + 4 + ln > len(buf):
                    break   # torn tail — re-read next round
                region.series.add_encoded(buf[pos + 4: pos + 4 + ln])
                pos += 4 + ln
        self._series_offsets[region.region_id] = off + pos

    def catchup(self) -> int:
        """Apply leader WAL entries above `applied_seq`; returns rows applied.
        (reference handle_catchup: replay WAL from flushed entry id)."""
        if self.role != "follower":
            return 0
        regions = {}
        for st in self.engine.tables.values():
            if self.tables is not None and st.schema.name not in self.tables:
                continue
            for r in st.regions:
                regions[r.region_id] = r
        for rid, region in regions.items():
            self._sync_series(
                region, os.path.join(self.leader_dir, "region", str(rid)))
        wal_dir = os.path.join(self.leader_dir, "wal")
        if not os.path.isdir(wal_dir):
            return 0
        applied = 0
        max_seq = self.applied_seq
        for seg in sorted(f for f in os.listdir(wal_dir) if f.endswith(".wal")):
            # Skip by PER-REGION high-water (region.last_seq), not a global
            # seq: with sharded WALs the shard files become durable in
            # arbitrary cross-shard order, so a global mark could skip a
            # lower-seq entry that appeared later. A region's entries stay
            # within one shard and are seq-ascending, so per-region marks
            # are exact.
            for rid, seq, payload in _native.wal_read_segment(
                    os.path.join(wal_dir, seg)):
                region = regions.get(rid)
                max_seq = max(max_seq, seq)
                if region is None:
                    continue
                if seq <= region.last_seq:
                    continue
                series, ts, fields, fnames, new_series, str_cols = \
                    decode_batch(payload)
                for _code, pk in new_series:
                    region.series.add_encoded(pk)
                if fnames != region.field_names:
                    with self.engine._ddl_lock:
                        missing = [fn for fn in fnames
                                   if fn not in region.field_names]
                        st = self.engine.tables[
                            region.schema.name] if hasattr(region, "schema") else None
                        for r2 in (st.regions if st else [region]):
                            r2.ensure_fields(missing)
                    fmap = {fn: i for i, fn in enumerate(fnames)}
                    out = np.full((len(region.field_names), fields.shape[1]),
                                  np.nan)
                    for i, fn in enumerate(region.field_names):
                        if fn in fmap:
                            out[i] = fields[fmap[fn]]
                    fields = out
                region.append(series, ts, fields, seq, str_cols or None)
                applied += len(ts)
        self.applied_seq = max_seq
        return applied

    # ------------------------------------------------------------ failover

    def promote(self) -> None:
        """Follower → leader (failover): final catchup, then flip the role.
        The data is already resident in this GPU's memtables — promotion
        moves no bytes (reference: Follower→StagingLeader→Leader ladder)."""
        self.catchup()
        self.role = "leader"
