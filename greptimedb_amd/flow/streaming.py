"""Streaming flow mode: incremental dataflow state (no source re-scan).

Reference parity: src/flow's streaming engine — a Plan rendered to an
executable dataflow with per-operator incremental state
(src/flow/src/compute/render.rs:15, compute/state.rs), fed by frontend
insert mirroring (src/operator/src/insert.rs:1384 FlowMirrorTask). MI355X
redesign: the mirror hook fires inside the engine write path with the
already-parsed columnar batch; the reduce operator keys state by
(region, series-code, bucket) — dense integers, vectorized with numpy —
and resolves tag values only when flushing to the sink table.

Supported plan shape (everything else falls back to the batching mode, as
the reference's FlowDualEngine routes): SELECT <tags...>,
date_bin/date_trunc(ts) [alias], agg(field)... FROM src [WHERE numeric/ts
predicate] GROUP BY <tags..., bucket>.
"""

from __future__ import annotations

import threading

import numpy as np

from greptimedb_amd.query import ast
from greptimedb_amd.query.parser import parse_sql
from greptimedb_amd.utils.timeutil import trunc_unit_ms

AGGS = {"count", "sum", "min", "max", "avg", "mean"}


class StreamingPlan:
    """Validated incremental plan: group tags + bucket + aggregate list."""

    def __init__(self, sel: ast.Select, schema):
        self.source = sel.table
        tag_names = {c.name for c in schema.tag_columns}
        ts_name = schema.time_index.name
        self.ts_name = ts_name
        self.bucket_ms = None
        self.bucket_name = "ts"
        self.group_tags: list[str] = []
        self.aggs: list[tuple[str, str, str]] = []  # (fn, field, out_name)
        self.out_cols: list[tuple[str, str]] = []   # (kind, name) in order
        alias_of: dict[int, str] = {}
        # WHERE: only ts-range and field comparisons supported incrementally
        self.where = sel.where
        if sel.having or sel.joins or sel.limit is not None or sel.order_by:
            raise ValueError("unsupported clauses for streaming")
        for e, alias in sel.projections:
            if isinstance(e, ast.Col) and e.name in tag_names:
                self.group_tags.append(e.name)
                self.out_cols.append(("tag", alias or e.name))
            elif isinstance(e, ast.Func) and e.name in ("date_bin", "date_trunc"):
                if self.bucket_ms is not None:
                    raise ValueError("one bucket expr only")
                self.bucket_ms = _bucket_ms_of(e)
                self.bucket_name = alias or "ts"
                self.out_cols.append(("bucket", self.bucket_name))
            elif isinstance(e, ast.Func) and e.name.lower() in AGGS:
                fn = e.name.lower()
                default_name = None
                if fn == "count" and (not e.args or isinstance(e.args[0], ast.Star)):
                    field = ts_name  # count(*) — any non-null column
                    fn = "count*"
                    default_name = "count(*)"
                elif len(e.args) == 1 and isinstance(e.args[0], ast.Col):
                    field = e.args[0].name
                else:
                    raise ValueError("aggregate arg must be a column")
                name = alias or default_name or f"{e.name}({field})"
                self.aggs.append((fn, field, name))
                self.out_cols.append(("agg", name))
            else:
                raise ValueError(f"unsupported projection {e}")
        if not self.aggs:
            raise ValueError("streaming flow needs at least one aggregate")
        # GROUP BY must cover exactly the tags + bucket we emit
        want = len(self.group_tags) + (1 if self.bucket_ms else 0)
        if len(sel.group_by) != want:
            raise ValueError("GROUP BY must match projected keys")


def _bucket_ms_of(e: ast.Func) -> int:
    if e.name == "date_trunc":
        unit = e.args[0].value
        return trunc_unit_ms(str(unit))
    a0 = e.args[0]
    if isinstance(a0, ast.Interval):
        return a0.ms
    if isinstance(a0, ast.Lit):
        from greptimedb_amd.query.parser import parse_interval_text
        return parse_interval_text(str(a0.value))
    raise ValueError("date_bin needs an interval literal")


class StreamingFlowTask:
    """One flow's incremental reduce state (compute/state.rs analog)."""

    def __init__(self, name: str, sink: str, select_sql: str, engine,
                 expire_after_s: int | None = None):
        self.name = name
        self.sink = sink
        self.select_sql = select_sql
        self.engine = engine
        self.expire_after_s = expire_after_s
        sel = parse_sql(select_sql)
        if not isinstance(sel, ast.Select) or not isinstance(sel.table, str):
            raise ValueError("flow query must be SELECT ... FROM table")
        schema = engine.table(sel.table).schema
        self.plan = StreamingPlan(sel, schema)
        self.lock = threading.Lock()
        # state[(region_id, code, bucket)] = [per-agg accumulators]
        #   count*: n ; sum/avg: (s, n) ; min: m ; max: m ; count: n
        self.state: dict = {}
        self._dirty = False

    # ------------------------------------------------------------ ingest
    def on_write(self, st, region, codes: np.ndarray, ts_ms: np.ndarray,
                 fields: np.ndarray, field_names: list[str]):
        """Mirror one columnar write batch into the reduce state."""
        plan = self.plan
        n = len(ts_ms)
        if n == 0:
            return
        keep = np.ones(n, dtype=bool)
        if plan.where is not None:
            keep = self._where_mask(plan.where, ts_ms, fields, field_names, n)
            if not keep.any():
                return
        idx = np.flatnonzero(keep)
        bucket_ms = plan.bucket_ms or 1
        buckets = (ts_ms[idx] // bucket_ms) * bucket_ms if plan.bucket_ms \
            else np.zeros(len(idx), dtype=np.int64)
        cset = codes[idx]
        fpos = {fn: i for i, fn in enumerate(field_names)}
        cols = {}
        for fn_, field, _name in plan.aggs:
            if field not in cols:
                j = fpos.get(field)
                cols[field] = fields[j][idx] if j is not None else \
                    np.full(len(idx), np.nan)
        with self.lock:
            rid = region.region_id
            state = self.state
            for k in range(len(idx)):
                key = (rid, int(cset[k]), int(buckets[k]))
                accs = state.get(key)
                if accs is None:
                    accs = state[key] = [self._init_acc(fn)
                                         for fn, _f, _n2 in plan.aggs]
                for a_i, (fn, field, _nm) in enumerate(plan.aggs):
                    v = cols[field][k] if fn != "count*" else 1.0
                    accs[a_i] = self._step(fn, accs[a_i], v)
            self._dirty = True

    @staticmethod
    def _init_acc(fn):
        if fn in ("sum", "avg", "mean"):
            return (0.0, 0)
        if fn in ("count", "count*"):
            return 0
        return None  # min/max

    @staticmethod
    def _step(fn, acc, v):
        if fn == "count*":
            return acc + 1
        if v is None or (isinstance(v, float) and np.isnan(v)):
            return acc
        if fn == "count":
            return acc + 1
        if fn in ("sum", "avg", "mean"):
            return (acc[0] + v, acc[1] + 1)
        if fn == "min":
            return v if acc is None else min(acc, v)
        return v if acc is None else max(acc, v)

    def _where_mask(self, e, ts_ms, fields, field_names, n) -> np.ndarray:
        fpos = {fn: i for i, fn in enumerate(field_names)}

        def val(x):
            if isinstance(x, ast.Col):
                if x.name == self.plan.ts_name:
                    return ts_ms.astype(np.float64)
                j = fpos.get(x.name)
                if j is None:
                    raise ValueError(f"streaming WHERE: unknown column {x.name}")
                return fields[j]
            if isinstance(x, ast.Lit):
                return float(x.value)
            raise ValueError("streaming WHERE supports col OP literal")

        if isinstance(e, ast.BinOp):
            if e.op == "and":
                return self._where_mask(e.left, ts_ms, fields, field_names, n) & \
                    self._where_mask(e.right, ts_ms, fields, field_names, n)
            if e.op == "or":
                return self._where_mask(e.left, ts_ms, fields, field_names, n) | \
                    self._where_mask(e.right, ts_ms, fields, field_names, n)
            l, r = val(e.left), val(e.right)
            return np.asarray({"=": l == r, "!=": l != r, "<": l < r,
                               "<=": l <= r, ">": l > r, ">=": l >= r}[e.op])
        raise ValueError("unsupported streaming WHERE")

    # ------------------------------------------------------------ flush
    def flush(self) -> int:
        """Emit the current state into the sink table (upsert; merge-mode
        sink overwrites per (tags, bucket)). Returns rows written."""
        with self.lock:
            if not self._dirty:
                return 0
            items = list(self.state.items())
            self._dirty = False
        if not items:
            return 0
        plan = self.plan
        engine = self.engine
        src_st = engine.table(plan.source)
        regions_by_id = {r.region_id: r for r in src_st.regions}
        rows = []
        import time as _time
        floor = None
        if self.expire_after_s is not None:
            floor = int(_time.time() * 1000) - self.expire_after_s * 1000
        for (rid, code, bucket), accs in items:
            if floor is not None and plan.bucket_ms and bucket < floor:
                continue
            region = regions_by_id.get(rid)
            tags_all = region.series.tag_values[code] if region else ()
            tag_map = dict(zip([c.name for c in src_st.schema.tag_columns],
                               tags_all))
            row = {}
            for tg in plan.group_tags:
                row[tg] = tag_map.get(tg)
            row["__bucket"] = bucket
            for a_i, (fn, _f, name) in enumerate(plan.aggs):
                acc = accs[a_i]
                if fn in ("sum", "avg", "mean"):
                    s, c = acc
                    row[name] = (np.nan if c == 0 else
                                 (s if fn == "sum" else s / c))
                elif fn in ("count", "count*"):
                    row[name] = float(acc)
                else:
                    row[name] = np.nan if acc is None else acc
            rows.append(row)
        if not rows:
            return 0
        return self._upsert_sink(rows)

    def _upsert_sink(self, rows: list[dict]) -> int:
        from greptimedb_amd.models.schema import (ColumnSchema, DataType,
                                                  SemanticType, TableSchema)
        engine = self.engine
        plan = self.plan
        agg_names = [nm for _fn, _f, nm in plan.aggs]
        try:
            st = engine.table(self.sink)
        except Exception:
            cols, cid = [], 0
            for tg in plan.group_tags:
                cols.append(ColumnSchema(tg, DataType.STRING,
                                         SemanticType.TAG, cid)); cid += 1
            cols.append(ColumnSchema(plan.bucket_name, DataType.TIMESTAMP_MS,
                                     SemanticType.TIMESTAMP, cid)); cid += 1
            for nm in agg_names:
                cols.append(ColumnSchema(nm, DataType.FLOAT64,
                                         SemanticType.FIELD, cid)); cid += 1
            st = engine.create_table(
                TableSchema(name=self.sink, columns=cols,
                            primary_key=list(plan.group_tags)),
                append_mode=False, if_not_exists=True)
        new_f = [nm for nm in agg_names if nm not in st.regions[0].field_names]
        if new_f:
            for reg in st.regions:
                reg.ensure_fields(new_f)
        sink_fnames = st.regions[0].field_names
        n = len(rows)
        ts_vals = np.array([r["__bucket"] for r in rows], dtype=np.int64)
        codes = np.empty(n, dtype=np.int32)
        rows_by_region: dict[int, list[int]] = {}
        for i, r in enumerate(rows):
            tags = tuple(r.get(tg) for tg in plan.group_tags)
            ridx = engine.region_of_tags(st, tags)
            codes[i] = st.regions[ridx].register_series(tags)
            rows_by_region.setdefault(ridx, []).append(i)
        for ridx, idxs in rows_by_region.items():
            ia = np.array(idxs)
            fmat = np.full((len(sink_fnames), len(ia)), np.nan)
            for j, fn in enumerate(sink_fnames):
                if fn in agg_names:
                    fmat[j] = [rows[i][fn] for i in ia]
            engine.write_region(st, ridx, codes[ia], ts_vals[ia], fmat, [])
        engine.commit_wal()
        return n
