"""Flow engine: continuous aggregation ("batching mode").

Reference parity: src/flow — specifically the batching mode
(batching_mode.rs: time-window-aware SQL re-query on new data), which is
the production-recommended mode; the streaming dataflow mode collapses into
the same mechanism here because re-querying a dirty window hits
device-resident columns (a windowed GPU aggregate is cheaper than
maintaining incremental operator state).

A flow = source SELECT (with a time-bucket GROUP BY) + sink table. Writes
to the source mark [min_ts, max_ts] dirty (engine write listener); tick()
re-evaluates the query restricted to dirty windows and upserts the result
into the sink (merge-mode sink → last-wins overwrite per (tags, bucket)).
"""

from __future__ import annotations

import threading

import numpy as np

from greptimedb_amd.models.schema import ColumnSchema, DataType, SemanticType, TableSchema
from greptimedb_amd.query import ast
from greptimedb_amd.query.parser import parse_sql
from greptimedb_amd.utils.errors import InvalidArguments


class FlowTask:
    def __init__(self, name: str, sink: str, select_sql: str,
                 expire_after_s: int | None = None):
        self.name = name
        self.sink = sink
        self.select_sql = select_sql
        self.expire_after_s = expire_after_s
        self.select = parse_sql(select_sql)
        if not isinstance(self.select, ast.Select) or self.select.table is None:
            raise InvalidArguments("flow query must be a SELECT ... FROM table")
        self.source = self.select.table
        self.dirty_lo: int | None = None
        self.dirty_hi: int | None = None
        self.lock = threading.Lock()

    def mark_dirty(self, lo: int, hi: int):
        with self.lock:
            self.dirty_lo = lo if self.dirty_lo is None else min(self.dirty_lo, lo)
            self.dirty_hi = hi if self.dirty_hi is None else max(self.dirty_hi, hi)

    def take_dirty(self):
        with self.lock:
            d = (self.dirty_lo, self.dirty_hi)
            self.dirty_lo = self.dirty_hi = None
            return d


class FlowEngine:
    """Dual-mode flow engine (reference FlowDualEngine, src/flow/src/
    engine.rs:15): flows whose plan shape supports incremental reduction
    run STREAMING (flow/streaming.py — per-write mirror into operator
    state, no source re-scan); everything else runs BATCHING (dirty-window
    re-query below)."""

    def __init__(self, engine, executor):
        self.engine = engine
        self.executor = executor
        self.flows: dict[str, FlowTask] = {}
        engine.write_listeners.append(self._on_write)
        engine.mirror_listeners.append(self._on_mirror)

    def _on_write(self, table: str, lo: int, hi: int, n: int):
        for f in self.flows.values():
            if isinstance(f, FlowTask) and f.source == table:
                f.mark_dirty(lo, hi)

    def _on_mirror(self, st, region, codes, ts_ms, fields, field_names):
        from greptimedb_amd.flow.streaming import StreamingFlowTask
        name = st.schema.name
        for f in self.flows.values():
            if isinstance(f, StreamingFlowTask) and f.plan.source == name:
                f.on_write(st, region, codes, ts_ms, fields, field_names)

    def create_flow(self, name: str, sink: str, select_sql: str,
                    if_not_exists: bool = False,
                    expire_after_s: int | None = None):
        if name in self.flows:
            if if_not_exists:
                return
            raise InvalidArguments(f"flow {name} exists")
        from greptimedb_amd.flow.streaming import StreamingFlowTask
        try:
            task = StreamingFlowTask(name, sink, select_sql, self.engine,
                                     expire_after_s=expire_after_s)
        except Exception:
            # shape not incrementally reducible (or source not created
            # yet) → batching mode, like the reference's dual-engine route
            task = FlowTask(name, sink, select_sql, expire_after_s)
        self.flows[name] = task

    def mode_of(self, name: str) -> str:
        f = self.flows.get(name)
        return "batching" if isinstance(f, FlowTask) else "streaming"

    def drop_flow(self, name: str):
        self.flows.pop(name, None)

    def tick(self) -> dict[str, int]:
        """Streaming flows flush their incremental state; batching flows
        re-evaluate dirty windows (reference: batching-mode task tick)."""
        out = {}
        import time as _time
        from greptimedb_amd.flow.streaming import StreamingFlowTask
        for f in self.flows.values():
            if isinstance(f, StreamingFlowTask):
                r = f.flush()
                if r:
                    out[f.name] = r
        for f in self.flows.values():
            if isinstance(f, StreamingFlowTask):
                continue
            lo, hi = f.take_dirty()
            if lo is None:
                continue
            if f.expire_after_s is not None:
                # EXPIRE AFTER: data older than the TTL never re-aggregates
                # (reference: batching-mode expire_after window clamp)
                floor = int(_time.time() * 1000) - f.expire_after_s * 1000
                lo = max(lo, floor)
                if lo > hi:
                    out[f.name] = 0
                    continue
            out[f.name] = self._run_flow(f, lo, hi + 1)
        return out

    def _run_flow(self, f: FlowTask, ts_lo: int, ts_hi: int) -> int:
        sel = parse_sql(f.select_sql)  # fresh AST (executor may rewrite)
        ts_name = self.engine.table(f.source).schema.time_index.name
        bound = ast.BinOp("and",
                          ast.BinOp(">=", ast.Col(ts_name), ast.Lit(int(ts_lo))),
                          ast.BinOp("<", ast.Col(ts_name), ast.Lit(int(ts_hi))))
        sel.where = bound if sel.where is None else ast.BinOp("and", sel.where, bound)
        r = self.executor.execute_stmt(sel)
        if len(r) == 0:
            return 0
        # sink schema from result: ts-kind column → time index, object
        # columns → tags, numeric → fields
        ts_col = next((i for i, k in enumerate(r.kinds) if k == "ts"), None)
        tag_idx = [i for i, c in enumerate(r.columns)
                   if i != ts_col and len(c) and isinstance(c[0], str)]
        field_idx = [i for i in range(len(r.columns))
                     if i != ts_col and i not in tag_idx]
        try:
            st = self.engine.table(f.sink)
        except Exception:
            cols = []
            cid = 0
            for i in tag_idx:
                cols.append(ColumnSchema(r.names[i], DataType.STRING,
                                         SemanticType.TAG, cid)); cid += 1
            cols.append(ColumnSchema("ts", DataType.TIMESTAMP_MS,
                                     SemanticType.TIMESTAMP, cid)); cid += 1
            for i in field_idx:
                cols.append(ColumnSchema(r.names[i], DataType.FLOAT64,
                                         SemanticType.FIELD, cid)); cid += 1
            schema = TableSchema(name=f.sink, columns=cols,
                                 primary_key=[r.names[i] for i in tag_idx])
            st = self.engine.create_table(schema, append_mode=False,
                                          if_not_exists=True)
        # upsert rows (merge-mode sink: same (tags, ts) overwrites)
        from greptimedb_amd.engine import pk_codec
        from greptimedb_amd.engine.series import tsid_hash
        n = len(r)
        ts_vals = (np.asarray(r.columns[ts_col], dtype=np.int64) if ts_col is not None
                   else np.full(n, int(ts_hi - 1), dtype=np.int64))
        fnames = [r.names[i] for i in field_idx]
        new_f = [fn for fn in fnames if fn not in st.regions[0].field_names]
        if new_f:
            for reg in st.regions:
                reg.ensure_fields(new_f)
        sink_fnames = st.regions[0].field_names
        rows_by_region: dict[int, list[int]] = {}
        codes = np.empty(n, dtype=np.int32)
        for i in range(n):
            tags = tuple(str(r.columns[j][i]) for j in tag_idx)
            pk = pk_codec.encode_pk(tags)
            ridx = tsid_hash(pk) % len(st.regions)
            codes[i] = st.regions[ridx].register_series(tags)
            rows_by_region.setdefault(ridx, []).append(i)
        for ridx, rows in rows_by_region.items():
            rows_a = np.array(rows)
            fmat = np.full((len(sink_fnames), len(rows_a)), np.nan)
            for j, fn in enumerate(sink_fnames):
                if fn in fnames:
                    src = np.asarray(r.columns[field_idx[fnames.index(fn)]],
                                     dtype=np.float64)
                    fmat[j] = src[rows_a]
            self.engine.write_region(st, ridx, codes[rows_a], ts_vals[rows_a],
                                     fmat, [])
        self.engine.commit_wal()
        return n
