"""Region: one shard of a table — GPU memtable + SSTs + manifest.

Reference parity: src/mito2 region (region/version.rs VersionControl,
flush.rs, read/scan_region.rs). MI355X redesign: all row data for a region
lives on ONE GPU (the region↔GPU mapping is the P1 sharding axis,
SURVEY.md §2.8); flush sorts/dedups on device, writes the parquet SST from
a host copy, and keeps the sorted device tensors as the scan cache.
"""

from __future__ import annotations

import os
import struct
import threading

import numpy as np
import torch

from greptimedb_amd.engine import sst as sst_mod
from greptimedb_amd.engine.manifest import Manifest
from greptimedb_amd.engine.memtable import Memtable
from greptimedb_amd.engine.series import SeriesIndex
from greptimedb_amd.models.schema import TableSchema
from greptimedb_amd.ops import dedup_mark_last


class ScanSource:
    """One scan input: device column set + field-name → row mapping.
    str_cols: host string columns ({name: list|ndarray}), text_probe:
    callable(col, terms) → bool mask (fulltext MATCHES)."""

    __slots__ = ("ts", "series", "fields", "n", "field_pos", "sorted",
                 "str_cols", "text_probe")

    def __init__(self, ts, series, fields, n, field_pos, sorted,
                 str_cols=None, text_probe=None):
        self.ts = ts
        self.series = series
        self.fields = fields
        self.n = n
        self.field_pos = field_pos
        self.sorted = sorted
        self.str_cols = str_cols or {}
        self.text_probe = text_probe


class Region:
    def __init__(self, region_id: int, schema: TableSchema, dir: str,
                 device: str = "cpu", append_mode: bool = False):
        self.region_id = region_id
        self.schema = schema
        self.dir = dir
        self.device = device
        self.append_mode = append_mode
        self.field_names = [c.name for c in schema.field_columns
                            if not c.dtype.is_string_like]
        # string fields (log columns): host-side values; fulltext index only
        # where the column opts in (schema fulltext flag)
        self.str_field_names = [c.name for c in schema.field_columns
                                if c.dtype.is_string_like]
        from greptimedb_amd.engine.fulltext import FulltextColumn
        self.text_cols: dict[str, FulltextColumn] = {
            c.name: FulltextColumn() for c in schema.field_columns
            if c.dtype.is_string_like and c.fulltext}
        self.series = SeriesIndex([c.name for c in schema.tag_columns])
        os.makedirs(os.path.join(dir, "sst"), exist_ok=True)
        self.manifest = Manifest(os.path.join(dir, "manifest"))
        self.sst_cache: dict[str, sst_mod.SstBatch] = {}
        self.flushed_seq = self.manifest.flushed_seq
        self.last_seq = self.flushed_seq
        self.lock = threading.Lock()
        # Per-row sequence space (independent of WAL entry seq): every
        # appended row gets a monotonically increasing u64, stored in the
        # SST __sequence column so overlapping files order correctly under
        # a reference-style merge (mito2 region_write_ctx.rs publishes a
        # committed_sequence the same way).
        self.row_seq = 0          # next row sequence to assign
        self.mem_base = 0         # row seq of active memtable's row 0
        # Memtables swapped out by an in-progress flush stay visible to
        # scans here until their SstBatch is published (mito2 keeps
        # immutable memtables in Version until the flush edit lands).
        self.flushing: list = []  # [(memtable, n_rows, mem_postings, base)]
        self._flush_lock = threading.Lock()
        # Role fencing (reference store-api RegionRole Leader/Follower/
        # DowngradingLeader): a downgraded region rejects writes while a
        # migration moves ownership.
        self.writable = True
        # Series code assignment must be stable across restarts (GPU columns
        # and SST caches store dense codes): an append-only series log is the
        # authoritative code order. Loaded BEFORE SSTs.
        self._series_log_path = os.path.join(dir, "series.log")
        self._load_series_log()
        self._series_log = open(self._series_log_path, "ab")
        self._load_strcols_meta()   # restore field/str/fulltext layout
        # memtable AFTER the meta sidecar: auto-ALTERed field columns (the
        # influx auto-create path evolves the schema without DDL) must be
        # known before sizing the field matrix and reading SSTs
        self.memtable = Memtable(len(self.field_names), device=device)
        self._load_ssts()
        self.mem_base = self.row_seq

    def _load_series_log(self):
        if not os.path.exists(self._series_log_path):
            return
        with open(self._series_log_path, "rb") as f:
            buf = f.read()
        off = 0
        while off + 4 <= len(buf):
            (ln,) = struct.unpack_from("<I", buf, off)
            if off + 4 + ln > len(buf):
                break  # torn tail
            self.series.add_encoded(buf[off + 4: off + 4 + ln])
            off += 4 + ln

    def register_series(self, tags: tuple) -> int:
        """Get-or-create a local series code; new codes are logged durably."""
        with self.lock:
            prev = len(self.series)
            code = self.series.get_or_create(tags)
            if code >= prev:
                pk = self.series.pks[code]
                self._series_log.write(struct.pack("<I", len(pk)) + pk)
                self._series_log.flush()
            return code

    def register_series_labels(self, labels: dict) -> int:
        """Sparse/metric-engine mode: dynamic label-set series."""
        with self.lock:
            prev = len(self.series)
            code = self.series.get_or_create_labels(labels)
            if code >= prev:
                pk = self.series.pks[code]
                self._series_log.write(struct.pack("<I", len(pk)) + pk)
                self._series_log.flush()
            return code

    def register_series_bulk(self, labels_list) -> np.ndarray:
        """Bulk registration (fixture/bulk-ingest path): one series-log
        flush for the whole batch. Items are dicts (sparse/metric-engine
        mode) or tuples (dense tag order)."""
        prev = len(self.series)
        codes = np.empty(len(labels_list), dtype=np.int32)
        for i, labels in enumerate(labels_list):
            if isinstance(labels, dict):
                codes[i] = self.series.get_or_create_labels(labels)
            else:
                codes[i] = self.series.get_or_create(labels)
        buf = bytearray()
        for code in range(prev, len(self.series)):
            pk = self.series.pks[code]
            buf += struct.pack("<I", len(pk)) + pk
        if buf:
            self._series_log.write(bytes(buf))
            self._series_log.flush()
        return codes

    # ---------------------------------------------------------------- open

    def _load_ssts(self):
        """Open path: load manifest SSTs into the device cache. On GPU the
        numeric columns decode through the K11 path (host thrift+zstd page
        parse → device RLE/dict expand, engine/pagedec.py) so pages land in
        HBM without a CPU Arrow materialization; string columns and the pk
        dictionary still go through pyarrow."""
        for fid, meta in self.manifest.files.items():
            path = os.path.join(self.dir, "sst", f"{fid}.parquet")
            if not os.path.exists(path):
                continue
            dev = self.device
            gpu_cols = None
            if str(dev).startswith("cuda") and not self.str_field_names:
                try:
                    gpu_cols = self._load_sst_gpu(path)
                except Exception:
                    gpu_cols = None   # unsupported encoding → CPU reader
            if gpu_cols is not None:
                t_ts, t_f, t_seq, codes_t = gpu_cols
                t_se = codes_t
                str_cols = {}
                ts = t_ts  # only len used below via tensors
                seq = t_seq
            else:
                dict_values, indices, ts, fields, seq, str_cols = \
                    sst_mod.read_sst(path, self.schema, self.field_names)
                remap = np.array([self.series.add_encoded(pk)
                                  for pk in dict_values], dtype=np.int32)
                codes = remap[indices]
                t_ts = torch.as_tensor(ts).to(dev)
                t_se = torch.as_tensor(codes).to(dev)
                t_f = torch.as_tensor(np.ascontiguousarray(fields)).to(dev)
                t_seq = torch.as_tensor(seq).to(dev)
            # re-sort by (series, ts, seq): codes may differ from write-time order
            ord1 = torch.argsort(t_seq, stable=True)
            ord2 = torch.argsort(t_ts[ord1], stable=True)
            perm = ord1[ord2]
            ord3 = torch.argsort(t_se[perm], stable=True)
            perm = perm[ord3]
            batch = sst_mod.SstBatch(
                t_ts[perm].contiguous(), t_se[perm].contiguous(),
                t_f[:, perm].contiguous(), t_seq[perm].contiguous(),
                int(meta["min_ts"]), int(meta["max_ts"]),
                list(self.field_names))
            if len(seq):
                self.row_seq = max(self.row_seq, int(seq.max()) + 1)
            if str_cols:
                perm_h = perm.cpu().numpy()
                # persisted sidecar (Puffin analog): postings load directly,
                # remapped through the open-time permutation — no doc
                # re-tokenization (reference: index blobs read from Puffin)
                loaded = None
                if any(name in self.text_cols for name in str_cols):
                    from greptimedb_amd.engine import ftindex
                    inv = np.empty(len(perm_h), dtype=np.int64)
                    inv[perm_h] = np.arange(len(perm_h), dtype=np.int64)
                    loaded = ftindex.load_sidecar(path, self.text_cols,
                                                  self.device, row_remap=inv)
                for name, vals in str_cols.items():
                    if name not in self.str_field_names:
                        self.str_field_names.append(name)
                    arr = vals[perm_h]
                    batch.str_cols[name] = arr
                    ft = self.text_cols.get(name)
                    if ft is not None:  # fulltext-indexed columns only
                        if loaded is not None and name in loaded:
                            batch.text_index[name] = loaded[name]
                        else:  # pre-sidecar SSTs: rebuild from raw strings
                            batch.text_index[name] = ft.build_segment(
                                list(arr), self.device)
            self.sst_cache[fid] = batch

    def _load_sst_gpu(self, path: str):
        """K11 open path: numeric columns decode straight into HBM; only
        the (small) pk dictionary goes through pyarrow."""
        import pyarrow as pa
        import pyarrow.parquet as pq

        from greptimedb_amd.engine import pagedec
        dev = self.device
        ts_name = self.schema.time_index.name
        md0 = pq.read_metadata(path).row_group(0)
        avail = {md0.column(i).path_in_schema
                 for i in range(md0.num_columns)}
        # host page parse + zstd release the GIL (csrc/pagedec.cpp) →
        # decode all numeric columns in parallel threads
        from concurrent.futures import ThreadPoolExecutor
        want = [ts_name, "__sequence"] + [fn for fn in self.field_names
                                          if fn in avail]
        with ThreadPoolExecutor(max_workers=min(8, max(len(want), 1))) as tp:
            got = dict(zip(want, tp.map(
                lambda c: pagedec.read_numeric_column(path, c, dev), want)))
        t_ts = got[ts_name]
        t_seq = got["__sequence"]
        n = t_ts.numel()
        fparts = [got[fn] if fn in got else
                  torch.full((n,), float("nan"), dtype=torch.float64,
                             device=dev)
                  for fn in self.field_names]
        t_f = torch.stack(fparts) if fparts else \
            torch.zeros((0, n), dtype=torch.float64, device=dev)
        pk = pq.read_table(path, columns=["__primary_key"]) \
            .column("__primary_key").combine_chunks()
        if isinstance(pk, pa.ChunkedArray):
            pk = pk.chunk(0)
        dict_values = [v.as_py() for v in pk.dictionary]
        indices = pk.indices.to_numpy(zero_copy_only=False).astype(np.int32)
        remap = np.array([self.series.add_encoded(b) for b in dict_values],
                         dtype=np.int32)
        codes_t = torch.as_tensor(remap).to(dev)[
            torch.as_tensor(indices).to(dev).long()]
        return t_ts, t_f, t_seq.to(torch.int64), codes_t

    # ---------------------------------------------------------------- write

    def append(self, series_codes: np.ndarray, ts_ms: np.ndarray,
               fields: np.ndarray, last_seq: int,
               str_fields: dict[str, list] | None = None):
        with self.lock:
            n = len(ts_ms)
            if str_fields:
                for name in str_fields:
                    if name not in self.str_field_names:
                        self.str_field_names.append(name)
            self.memtable.append(series_codes, ts_ms, fields, str_fields)
            # keep every text column's row numbering aligned with the memtable
            for name, ft in self.text_cols.items():
                vals = (str_fields or {}).get(name)
                ft.index_batch(vals if vals is not None else [None] * n)
            self.last_seq = max(self.last_seq, last_seq)
            self.row_seq += n

    def should_flush(self, limit_bytes: int) -> bool:
        return self.memtable.bytes_used >= limit_bytes

    def truncate(self):
        """Drop all data, keep schema + series registry (reference: mito2
        truncate — RegionMetaAction::Truncate clears the file list and the
        replay point; src/mito2/src/engine (truncate request path))."""
        with self._flush_lock:
            with self.lock:
                old = list(self.manifest.files)
                self.manifest.commit({"kind": "truncate",
                                      "flushed_seq": self.last_seq})
                self.flushed_seq = self.last_seq
                self.memtable = Memtable(len(self.field_names),
                                         device=self.device)
                self.mem_base = self.row_seq
                self.flushing.clear()
                self.sst_cache.clear()
                for ft in self.text_cols.values():
                    ft.reset_mem()
            for fid in old:
                for ext in (".parquet", ".ftidx"):
                    p = os.path.join(self.dir, "sst", f"{fid}{ext}")
                    try:
                        os.unlink(p)
                    except OSError:
                        pass

    def flush(self):
        """Sort + dedup the memtable on device, write an SST, swap memtable.

        The swapped-out memtable stays scannable via `self.flushing` until
        its SstBatch is published into sst_cache under the region lock, so
        concurrent scans never lose the flushing rows (mito2 keeps
        immutable memtables in Version the same way)."""
        with self._flush_lock:
            return self._flush_locked()

    def _flush_locked(self):
        with self.lock:
            mem = self.memtable
            if mem.len == 0:
                return None
            n = mem.len
            flush_seq = self.last_seq
            mem_base = self.mem_base
            flush_field_names = list(self.field_names[: mem.nf])
            self.memtable = Memtable(len(self.field_names), device=self.device,
                                     cap=max(mem.cap, 1 << 16))
            self.mem_base = self.row_seq
            mem_postings = {}
            for name, ft in self.text_cols.items():
                mem_postings[name] = ft.mem
                ft.reset_mem()  # new memtable rows start at 0
            entry = (mem, n, mem_postings, mem_base)
            self.flushing.append(entry)
        ts, se, fields, perm = mem.sorted_view()
        if not self.append_mode:
            keep = dedup_mark_last(se, ts)
            idx = keep.nonzero(as_tuple=True)[0]
            ts, se, fields, perm = ts[idx], se[idx], fields[:, idx], perm[idx]
        seq_t = perm.to(torch.int64) + mem_base
        ts_h = ts.cpu().numpy()
        se_h = se.cpu().numpy()
        f_h = fields.cpu().numpy()
        perm_h = perm.cpu().numpy()
        # string columns: reorder host-side, build device posting segments
        str_cols_sorted: dict[str, np.ndarray] = {}
        text_index = {}
        for name, col in mem.str_cols.items():
            arr = np.array(col[: n], dtype=object)[perm_h]
            str_cols_sorted[name] = arr
            ft = self.text_cols.get(name)
            if ft is not None:
                text_index[name] = ft.build_segment(list(arr), self.device)
        seq_h = perm_h.astype(np.int64) + mem_base
        fid = sst_mod.new_file_id()
        path = os.path.join(self.dir, "sst", f"{fid}.parquet")
        meta = sst_mod.write_sst(path, self.schema, self.series.pks,
                                 se_h, ts_h, f_h, seq_h, flush_field_names,
                                 str_cols=str_cols_sorted,
                                 region_id=self.region_id)
        if text_index:
            from greptimedb_amd.engine import ftindex
            ftindex.save_sidecar(path, text_index, self.text_cols)
        meta.seq_max = int(seq_h.max()) if len(seq_h) else 0
        self.manifest.commit({
            "kind": "edit",
            "files_to_add": [meta.to_dict()],
            "files_to_remove": [],
            "flushed_seq": flush_seq,
        })
        batch = sst_mod.SstBatch(
            ts.contiguous(), se.contiguous(), fields.contiguous(),
            seq_t.contiguous(), meta.min_ts, meta.max_ts, flush_field_names)
        batch.str_cols = str_cols_sorted
        batch.text_index = text_index
        with self.lock:
            self.sst_cache[fid] = batch
            self.flushed_seq = flush_seq
            self.flushing.remove(entry)
        return meta

    # ---------------------------------------------------------------- schema

    def ensure_str_fields(self, names: list[str], fulltext: bool = True):
        """Register string columns on this region (fulltext-indexed when
        requested — the log-pipeline default). The (name, fulltext) set is
        persisted in a region sidecar so an unflushed reopen (WAL-only
        data) restores the same index layout."""
        from greptimedb_amd.engine.fulltext import FulltextColumn
        with self.lock:
            changed = False
            for n in names:
                if n not in self.str_field_names:
                    self.str_field_names.append(n)
                    changed = True
                if fulltext and n not in self.text_cols:
                    ft = FulltextColumn()
                    mem = getattr(self, "memtable", None)  # None during open
                    ft.mem.n_rows = mem.len if mem is not None else 0
                    self.text_cols[n] = ft
                    changed = True
            if changed:
                self._save_strcols_meta()

    def _strcols_meta_path(self) -> str:
        return os.path.join(self.dir, "strcols.json")

    def _save_strcols_meta(self):
        """Region column-layout sidecar: dynamic numeric fields (influx
        auto-ALTER) + string/fulltext columns, so a reopen restores the
        exact layout even after the WAL entries that introduced the
        columns were purged (round-2 fix: previously only str cols)."""
        import json as _json
        tmp = self._strcols_meta_path() + ".tmp"
        with open(tmp, "w") as f:
            _json.dump({
                "version": 2,
                "fields": list(self.field_names),
                "strcols": {n: (n in self.text_cols)
                            for n in self.str_field_names},
            }, f)
        os.replace(tmp, self._strcols_meta_path())

    def _load_strcols_meta(self):
        import json as _json
        path = self._strcols_meta_path()
        if not os.path.exists(path):
            return
        with open(path) as f:
            meta = _json.load(f)
        if isinstance(meta, dict) and meta.get("version") == 2:
            strmap = meta.get("strcols", {})
            for fn in meta.get("fields", []):
                if fn not in self.field_names:
                    self.field_names.append(fn)
        else:   # round-1 format: {strcol: fulltext}
            strmap = meta
        plain = [n for n, ft in strmap.items() if not ft]
        fts = [n for n, ft in strmap.items() if ft]
        if plain:
            self.ensure_str_fields(plain, fulltext=False)
        if fts:
            self.ensure_str_fields(fts, fulltext=True)

    def ensure_fields(self, names: list[str]):
        """Auto-ALTER: add new field columns (reference insert.rs:562
        create_or_alter tables on demand). Layout persists in the region
        meta sidecar (reopen-safe once the WAL is purged)."""
        new = [n for n in names if n not in self.field_names]
        if not new:
            return
        with self.lock:
            self.field_names.extend(new)
            self.memtable.add_fields(len(new))
            self._save_strcols_meta()

    # ---------------------------------------------------------------- scan

    def scan_sources(self, ts_lo: int | None = None, ts_hi: int | None = None):
        """ScanSource list overlapping the time range, oldest → newest: SST
        cache batches (time-pruned, reference scan_region.rs:887), then
        memtables being flushed, then the active memtable. Everything is
        snapshotted under the region lock so a concurrent flush can neither
        hide rows nor mutate the dict mid-iteration."""
        out = []
        device = self.device
        with self.lock:
            batches = list(self.sst_cache.values())
            flushing = list(self.flushing)
            mem = self.memtable
            n = mem.len
        for batch in batches:
            if ts_lo is not None and batch.max_ts < ts_lo:
                continue
            if ts_hi is not None and batch.min_ts >= ts_hi:
                continue
            batch.ensure_decoded(device)   # K20 cold tier → HBM tensors

            def make_probe(b):
                def probe(col, terms):
                    ft = self.text_cols.get(col)
                    if ft is None:
                        return None
                    tids = ft.query_tids(terms)
                    if tids is None:
                        return torch.zeros(b.n, dtype=torch.bool, device=device)
                    seg = getattr(b, "text_index", {}).get(col)
                    if seg is None:
                        return torch.zeros(b.n, dtype=torch.bool, device=device)
                    return seg.probe(tids, device)
                return probe

            out.append(ScanSource(batch.ts, batch.series, batch.fields, batch.n,
                                  {fn: i for i, fn in enumerate(batch.field_names)},
                                  sorted=True,
                                  str_cols=getattr(batch, "str_cols", {}),
                                  text_probe=make_probe(batch)))
        for fmem, fn_rows, fpostings, _base in flushing:
            out.extend(self._mem_source(fmem, fn_rows, ts_lo, ts_hi,
                                        postings=fpostings))
        if n > 0:
            out.extend(self._mem_source(mem, n, ts_lo, ts_hi))
        return out

    def _mem_source(self, mem, n, ts_lo, ts_hi, postings=None):
        """Build the ScanSource for one (active or flushing) memtable;
        returns [] when the memtable's time range misses the scan range."""
        device = self.device
        if ts_lo is not None and mem.max_ts is not None and mem.max_ts < ts_lo:
            return []
        if ts_hi is not None and mem.min_ts is not None and mem.min_ts >= ts_hi:
            return []

        def mem_probe(col, terms, _n=n, _postings=postings):
            ft = self.text_cols.get(col)
            if ft is None:
                return None
            tids = ft.query_tids(terms)
            if tids is None:
                return torch.zeros(_n, dtype=torch.bool, device=device)
            pp = _postings.get(col) if _postings is not None else ft.mem
            if pp is None:
                return torch.zeros(_n, dtype=torch.bool, device=device)
            return pp.probe(tids, _n, device)

        return [ScanSource(mem.ts[:n], mem.series[:n], mem.fields, n,
                           {fn: i for i, fn in enumerate(self.field_names)},
                           sorted=False,
                           str_cols={k: v for k, v in mem.str_cols.items()},
                           text_probe=mem_probe)]

    @property
    def num_rows(self) -> int:
        with self.lock:
            return (self.memtable.len
                    + sum(n for _, n, _, _ in self.flushing)
                    + sum(b.n for b in self.sst_cache.values()))

    def time_range(self) -> tuple[int, int] | None:
        """(min_ts, max_ts) over memtables + SSTs, None if empty."""
        lo = hi = None
        with self.lock:
            batches = list(self.sst_cache.values())
            mems = [m for m, _, _, _ in self.flushing] + [self.memtable]
        for b in batches:
            lo = b.min_ts if lo is None else min(lo, b.min_ts)
            hi = b.max_ts if hi is None else max(hi, b.max_ts)
        for m in mems:
            if m.min_ts is not None:
                lo = m.min_ts if lo is None else min(lo, m.min_ts)
                hi = m.max_ts if hi is None else max(hi, m.max_ts)
        if lo is None:
            return None
        return lo, hi
