"""PromQL evaluator: AST → GPU window-evaluation over region shards.

Reference parity: src/promql extension plans (SeriesNormalize /
InstantManipulate / RangeManipulate / SeriesDivide — K7/K8/K9 in
SURVEY.md §2.7) + src/query/src/promql/planner.rs. MI355X redesign: instead
of streaming per-series manipulate operators, a selector is evaluated in ONE
kernel launch — window rows are gathered (slot, ts)-sorted per region,
concatenated (regions own disjoint slot ranges), and
ops.prom_range_eval computes the [S, T] sample matrix (instant lookback or
range function, Prometheus extrapolation semantics) in one pass.
Aggregations reduce the matrix on device; cross-rank combine all-reduces
group planes (parallel/dist.py).
"""

from __future__ import annotations

import math
import re as _re

import numpy as np
import torch

from greptimedb_amd.ops import dedup_mark_last, prom_range_eval
from greptimedb_amd.ops.cpu_ref import PROM_MODES
from greptimedb_amd.query.promql import ast
from greptimedb_amd.query.promql.parser import parse_promql
from greptimedb_amd.utils.errors import PlanQuery, TableNotFound

DEFAULT_LOOKBACK_S = 300

RANGE_FUNCS = {
    "rate": "rate", "increase": "increase", "delta": "delta",
    "idelta": "idelta", "irate": "irate", "deriv": "deriv",
    "avg_over_time": "avg_over_time", "sum_over_time": "sum_over_time",
    "min_over_time": "min_over_time", "max_over_time": "max_over_time",
    "count_over_time": "count_over_time", "last_over_time": "last_over_time",
    "stddev_over_time": "stddev_over_time", "stdvar_over_time": "stdvar_over_time",
    "present_over_time": "count_over_time",  # then >0 → 1
    "changes": "changes", "resets": "resets",
    "absent_over_time": "absent_over_time",
    "predict_linear": "predict_linear",
    "quantile_over_time": "quantile_over_time",
}

ELEMENTWISE = {
    "abs": torch.abs, "ceil": torch.ceil, "floor": torch.floor,
    "exp": torch.exp, "ln": torch.log, "log2": torch.log2,
    "log10": torch.log10, "sqrt": torch.sqrt, "sgn": torch.sgn,
    # trigonometric family (Prometheus experimental trig functions)
    "sin": torch.sin, "cos": torch.cos, "tan": torch.tan,
    "asin": torch.asin, "acos": torch.acos, "atan": torch.atan,
    "sinh": torch.sinh, "cosh": torch.cosh, "tanh": torch.tanh,
    "asinh": torch.asinh, "acosh": torch.acosh, "atanh": torch.atanh,
    "deg": lambda t: torch.rad2deg(t), "rad": lambda t: torch.deg2rad(t),
}

_TIME_FUNCS = {"minute", "hour", "day_of_week", "day_of_month",
               "day_of_year", "days_in_month", "month", "year"}


def _calendar_apply(f: str, secs: np.ndarray) -> np.ndarray:
    """UTC calendar component of epoch-seconds values (NaN preserved).
    Matches Prometheus: day_of_week 0=Sunday; 1-based day/month."""
    nanmask = np.isnan(secs)
    x = np.where(nanmask, 0.0, secs).astype(np.int64)
    if f == "minute":
        out = (x // 60) % 60
    elif f == "hour":
        out = (x // 3600) % 24
    elif f == "day_of_week":
        out = (x // 86400 + 4) % 7           # 1970-01-01 was a Thursday
    else:
        d = x.astype("datetime64[s]")
        Y = d.astype("datetime64[Y]")
        M = d.astype("datetime64[M]")
        D = d.astype("datetime64[D]")
        if f == "year":
            out = Y.astype(np.int64) + 1970
        elif f == "month":
            out = (M - Y).astype(np.int64) + 1
        elif f == "day_of_month":
            out = (D - M.astype("datetime64[D]")).astype(np.int64) + 1
        elif f == "day_of_year":
            out = (D - Y.astype("datetime64[D]")).astype(np.int64) + 1
        else:  # days_in_month
            out = ((M + 1).astype("datetime64[D]") -
                   M.astype("datetime64[D]")).astype(np.int64)
    return np.where(nanmask, np.nan, out.astype(np.float64))


class PromMatrix:
    """Evaluated vector: per-series labels + [S, T] value matrix (NaN =
    no sample) on the query grid.

    Labels are stored COLUMNAR (label name → np object array of length S) so
    10M-series selectors never build 10M Python dicts; `labels` materializes
    dicts lazily (cheap for post-aggregation results)."""

    def __init__(self, labels, values: torch.Tensor, grid: np.ndarray,
                 label_cols: dict | None = None, n_series: int | None = None):
        self._labels = labels          # list[dict] | None
        self.label_cols = label_cols   # {name: np object array [S]} | None
        self._n = n_series
        self.values = values
        self.grid = grid

    @property
    def S(self):
        if self._labels is not None:
            return len(self._labels)
        if self._n is not None:
            return self._n
        return int(self.values.shape[0])

    @property
    def labels(self) -> list:
        if self._labels is None:
            cols = self.label_cols or {}
            out = [{} for _ in range(self.S)]
            for n, (codes, values) in cols.items():
                for i in range(len(codes)):
                    c = codes[i]
                    if c >= 0:
                        out[i][n] = values[c]
            self._labels = out
        return self._labels

    @labels.setter
    def labels(self, v):
        self._labels = v
        self.label_cols = None

    def get_label_cols(self) -> dict:
        """Factorized columnar labels: {name: (value_id i32[S], values)}."""
        if self.label_cols is not None:
            return self.label_cols
        names = sorted({k for l in self.labels for k in l})
        out = {}
        for n in names:
            vmap: dict = {}
            codes = np.full(self.S, -1, dtype=np.int32)
            for i, l in enumerate(self.labels):
                v = l.get(n)
                if v is not None:
                    codes[i] = vmap.setdefault(v, len(vmap))
            out[n] = (codes, list(vmap))
        return out


class PromScalar:
    """Prometheus scalar type. `value` is a float (constant) or a 1-D
    float64 tensor of length T (per-step scalar — time(), scalar(v))."""

    def __init__(self, value, grid):
        self.value = value
        self.grid = grid


def _scalar_plane(sc: "PromScalar", T: int, device=None) -> torch.Tensor:
    """(1,T) tensor view of a scalar for vector contexts."""
    v = sc.value
    if isinstance(v, torch.Tensor):
        t = v.to(dtype=torch.float64)
        if device is not None:
            t = t.to(device)
        return t.reshape(1, -1).expand(1, T) if t.numel() != T else t.reshape(1, T)
    return torch.full((1, T), float(v), dtype=torch.float64, device=device)


class PromEvaluator:
    def __init__(self, engine, dist=None, lookback_s: int = DEFAULT_LOOKBACK_S):
        self.engine = engine
        self.dist = dist
        self.lookback_ms = lookback_s * 1000

    # ------------------------------------------------------------ entry

    def query_range(self, q: str, start_s: float, end_s: float, step_s: float) -> PromMatrix:
        expr = parse_promql(q) if isinstance(q, str) else q
        t0 = int(start_s * 1000)
        step = max(int(step_s * 1000), 1)
        T = int((int(end_s * 1000) - t0) // step) + 1
        grid = t0 + np.arange(T, dtype=np.int64) * step
        r = self._eval(expr, t0, step, T, grid)
        if isinstance(r, PromScalar):
            return PromMatrix([{}], _scalar_plane(r, T), grid)
        return r

    def query_instant(self, q: str, time_s: float) -> PromMatrix:
        return self.query_range(q, time_s, time_s, 1)

    # ------------------------------------------------------------ eval

    def _eval(self, e, t0, step, T, grid):
        if isinstance(e, ast.NumberLit):
            return PromScalar(e.value, grid)
        if isinstance(e, ast.StringLit):
            return PromScalar(e.value, grid)
        if isinstance(e, ast.Unary):
            r = self._eval(e.expr, t0, step, T, grid)
            if isinstance(r, PromScalar):
                return PromScalar(-r.value, grid)
            return PromMatrix(r.labels, -r.values, grid)
        if isinstance(e, ast.Selector):
            if e.range_s:
                raise PlanQuery("range vector must be wrapped in a function")
            return self._eval_selector(e, "instant", t0, step, T, grid,
                                       self.lookback_ms, 0.0)
        if isinstance(e, ast.Call):
            return self._eval_call(e, t0, step, T, grid)
        if isinstance(e, ast.Aggregate):
            return self._eval_aggregate(e, t0, step, T, grid)
        if isinstance(e, ast.BinOp):
            return self._eval_binop(e, t0, step, T, grid)
        raise PlanQuery(f"promql: unsupported node {type(e).__name__}")

    def _eval_call(self, e: ast.Call, t0, step, T, grid):
        f = e.func
        if f in RANGE_FUNCS:
            param = 0.0
            sel_idx = 0
            if f == "quantile_over_time":  # quantile_over_time(q, v[r])
                if len(e.args) != 2 or not isinstance(e.args[0], ast.NumberLit):
                    raise PlanQuery("quantile_over_time(scalar, v[r])")
                param = e.args[0].value
                sel_idx = 1
            if f == "predict_linear":      # predict_linear(v[r], t)
                if len(e.args) != 2 or not isinstance(e.args[1], ast.NumberLit):
                    raise PlanQuery("predict_linear(v[r], t)")
                param = e.args[1].value
            sel = e.args[sel_idx]
            if isinstance(sel, ast.Subquery):
                m = self._eval_subquery_range(sel, RANGE_FUNCS[f], t0, step, T,
                                              grid, param)
            elif not isinstance(sel, ast.Selector) or sel.range_s is None:
                raise PlanQuery(f"{f} needs a range vector argument")
            else:
                m = self._eval_selector(sel, RANGE_FUNCS[f], t0, step, T, grid,
                                        int(sel.range_s * 1000), param)
            if f == "present_over_time":
                v = m.values
                m = _matrix_map(m, torch.where(v > 0, torch.ones_like(v),
                                               torch.full_like(v, float("nan"))))
            if f == "absent_over_time" and m.S == 0:
                # no series matched at all → absent everywhere (Prometheus
                # emits a single 1-valued series)
                return PromMatrix([{}], torch.ones((1, T),
                                                   dtype=torch.float64), grid)
            if f in ("rate", "increase", "delta", "idelta", "irate", "deriv",
                     "predict_linear", "changes", "resets") or "_over_time" in f:
                m = _matrix_map(m, drop_name=True)
            return m
        if f == "predict_linear":  # handled above
            raise PlanQuery("unreachable")
        if f in ELEMENTWISE:
            m = self._eval(e.args[0], t0, step, T, grid)
            if isinstance(m, PromScalar):
                return PromScalar(float(ELEMENTWISE[f](torch.tensor(m.value))), grid)
            return _matrix_map(m, ELEMENTWISE[f](m.values), drop_name=True)
        if f == "round":
            m = self._eval(e.args[0], t0, step, T, grid)
            to = e.args[1].value if len(e.args) > 1 else 1.0
            v = torch.round(m.values / to) * to
            return _matrix_map(m, v, drop_name=True)
        if f in ("clamp", "clamp_min", "clamp_max"):
            m = self._eval(e.args[0], t0, step, T, grid)
            v = m.values
            if f == "clamp":
                v = v.clamp(e.args[1].value, e.args[2].value)
            elif f == "clamp_min":
                v = v.clamp_min(e.args[1].value)
            else:
                v = v.clamp_max(e.args[1].value)
            return _matrix_map(m, v, drop_name=True)
        if f == "scalar":
            m = self._eval(e.args[0], t0, step, T, grid)
            if isinstance(m, PromScalar):
                return m
            if m.S == 1:
                return PromScalar(m.values[0], grid)
            return PromScalar(torch.full((T,), float("nan"),
                                         dtype=torch.float64), grid)
        if f == "vector":
            m = self._eval(e.args[0], t0, step, T, grid)
            if isinstance(m, PromScalar):
                return PromMatrix([{}], _scalar_plane(m, T), grid)
            return m
        if f == "pi":
            return PromScalar(math.pi, grid)
        if f in ("sort_by_label", "sort_by_label_desc"):
            m = self._eval(e.args[0], t0, step, T, grid)
            keys = [a.value for a in e.args[1:]]
            order = sorted(range(m.S), key=lambda i: tuple(
                str(m.labels[i].get(k, "")) for k in keys),
                reverse=(f == "sort_by_label_desc"))
            import torch as _t
            oi = _t.as_tensor(order, device=m.values.device)
            return PromMatrix([m.labels[i] for i in order], m.values[oi],
                              grid)
        if f == "time":
            return PromScalar(torch.as_tensor(grid / 1000.0,
                                              dtype=torch.float64), grid)
        if f == "timestamp":
            m = self._eval(e.args[0], t0, step, T, grid)
            v = torch.where(torch.isnan(m.values),
                            m.values,
                            torch.as_tensor(grid[None, :] / 1000.0,
                                            dtype=torch.float64,
                                            device=m.values.device))
            return _matrix_map(m, v, drop_name=True)
        if f == "absent":
            m = self._eval(e.args[0], t0, step, T, grid)
            present = (~torch.isnan(m.values)).any(dim=0) if m.S else \
                torch.zeros(T, dtype=torch.bool)
            v = torch.where(present, torch.full((T,), float("nan"), dtype=torch.float64),
                            torch.ones(T, dtype=torch.float64))
            return PromMatrix([{}], v[None, :], grid)
        if f == "label_replace":
            m = self._eval(e.args[0], t0, step, T, grid)
            dst, repl, src, regex = (a.value for a in e.args[1:5])
            pat = _re.compile(regex)
            out = []
            for l in m.labels:
                l = dict(l)
                mm = pat.fullmatch(l.get(src, ""))
                if mm:
                    val = mm.expand(repl.replace("$", "\\"))
                    if val:
                        l[dst] = val
                    else:
                        l.pop(dst, None)
                out.append(l)
            return PromMatrix(out, m.values, grid)
        if f == "label_join":
            m = self._eval(e.args[0], t0, step, T, grid)
            dst = e.args[1].value
            sep = e.args[2].value
            srcs = [a.value for a in e.args[3:]]
            out = []
            for l in m.labels:
                l = dict(l)
                l[dst] = sep.join(l.get(s, "") for s in srcs)
                out.append(l)
            return PromMatrix(out, m.values, grid)
        if f == "histogram_quantile":
            return self._histogram_quantile(e, t0, step, T, grid)
        if f in ("sort", "sort_desc"):
            m = self._eval(e.args[0], t0, step, T, grid)
            if isinstance(m, PromScalar) or m.S <= 1:
                return m
            v = m.values[:, -1].cpu().numpy()
            key = np.where(np.isnan(v), np.inf if f == "sort" else -np.inf, v)
            order = np.argsort(-key if f == "sort_desc" else key, kind="stable")
            labels = [m.labels[i] for i in order]
            return PromMatrix(labels, m.values[torch.as_tensor(order.copy())],
                              grid)
        if f in _TIME_FUNCS:
            # minute(v=vector(time())) family — UTC calendar components
            if e.args:
                m = self._eval(e.args[0], t0, step, T, grid)
                if isinstance(m, PromScalar):
                    m = PromMatrix([{}], _scalar_plane(m, T), grid)
            else:
                m = PromMatrix([{}], torch.as_tensor(
                    grid[None, :] / 1000.0, dtype=torch.float64), grid)
            v = m.values.cpu().numpy()
            out = _calendar_apply(f, v)
            return _matrix_map(m, torch.as_tensor(out, dtype=torch.float64,
                                                  device=m.values.device),
                               drop_name=True)
        raise PlanQuery(f"promql: unsupported function {f}")

    def _histogram_quantile(self, e, t0, step, T, grid):
        q = e.args[0].value if isinstance(e.args[0], ast.NumberLit) else \
            self._eval(e.args[0], t0, step, T, grid).value
        m = self._eval(e.args[1], t0, step, T, grid)
        # group by labels minus 'le'
        groups: dict[tuple, list[tuple[float, int]]] = {}
        keys: dict[tuple, dict] = {}
        for i, l in enumerate(m.labels):
            le = l.get("le")
            if le is None:
                continue
            rest = tuple(sorted((k, v) for k, v in l.items() if k not in ("le", "__name__")))
            groups.setdefault(rest, []).append((float(le) if le != "+Inf" else math.inf, i))
            keys[rest] = {k: v for k, v in l.items() if k not in ("le", "__name__")}
        out_labels, rows = [], []
        vals = m.values
        for rest, buckets in groups.items():
            buckets.sort()
            les = [b[0] for b in buckets]
            idx = [b[1] for b in buckets]
            B = vals[idx]  # [nb, T]
            B = torch.cummax(B, dim=0).values  # enforce monotone
            total = B[-1]
            rank = q * total
            # first bucket with cum >= rank
            ge = (B >= rank[None, :])
            first = ge.float().argmax(dim=0)
            res = torch.full((T,), float("nan"), dtype=torch.float64)
            les_t = torch.tensor(les, dtype=torch.float64)
            for t in range(T):  # small T loops acceptable here
                tot = float(total[t])
                if not np.isfinite(tot) or tot == 0 or np.isnan(tot):
                    continue
                b = int(first[t])
                hi = les[b]
                lo = les[b - 1] if b > 0 else 0.0
                chi = float(B[b, t])
                clo = float(B[b - 1, t]) if b > 0 else 0.0
                if math.isinf(hi):
                    res[t] = les[b - 1] if b > 0 else float("nan")
                    continue
                r = float(rank[t])
                res[t] = lo + (hi - lo) * ((r - clo) / max(chi - clo, 1e-300))
            out_labels.append(keys[rest])
            rows.append(res)
        values = torch.stack(rows) if rows else torch.zeros((0, T), dtype=torch.float64)
        return PromMatrix(out_labels, values, grid)

    # ------------------------------------------------------------ selector

    def _resolve_table(self, sel: ast.Selector):
        name = sel.metric
        field = None
        for m in sel.matchers:
            if m.name == "__name__" and m.op == "=":
                name = m.value
            if m.name == "__field__" and m.op == "=":
                field = m.value
        if name is None:
            # bare label-matcher selector: evaluate against the metric-engine
            # physical table when present (all metrics share it)
            from greptimedb_amd.engine.promstore import PHYSICAL_TABLE, VALUE_FIELD
            phys = self.engine.tables.get(PHYSICAL_TABLE)
            if phys is not None:
                return phys, field or VALUE_FIELD
            raise PlanQuery("promql: metric name required")
        try:
            st = self.engine.table(name)
        except TableNotFound:
            # metric-engine physical table (remote-write metrics live there,
            # multiplexed by the __name__ label — engine/promstore.py)
            from greptimedb_amd.engine.promstore import PHYSICAL_TABLE, VALUE_FIELD
            phys = self.engine.tables.get(PHYSICAL_TABLE)
            if phys is not None:
                return phys, field or VALUE_FIELD
            # metric "table_field" convention (flat naming)
            for tname in self.engine.tables:
                if name.startswith(tname + "_"):
                    cand = name[len(tname) + 1:]
                    if cand in self.engine.tables[tname].regions[0].field_names:
                        return self.engine.tables[tname], cand
            return None, None
        if field is None:
            fns = st.regions[0].field_names
            preferred = [f for f in fns if f == "greptime_value"] or fns
            if len(preferred) != 1:
                raise PlanQuery(
                    f"promql: table {name} has {len(fns)} fields; add __field__ matcher")
            field = preferred[0]
        return st, field

    def _match_codes(self, region, sel: ast.Selector):
        """Codes matching all label matchers. Returns None (= all codes) or a
        sorted int64 array. Vectorized: regex/eq run over the label VALUE
        dictionary (small), then a boolean LUT gathers per-series — no
        per-series Python work (K13's host-side analog for labels)."""
        tag_names = region.series.tag_names
        n = len(region.series)
        mask = None  # np bool [n]
        matchers = list(sel.matchers)
        # metric-engine regions carry __name__ as an ordinary label
        if "__name__" in region.series.inverted:
            if sel.metric:
                matchers.append(ast.Matcher("__name__", "=", sel.metric))
        else:
            matchers = [m for m in matchers if m.name != "__name__"]
        for m in matchers:
            if m.name == "__field__":
                continue
            if m.name not in tag_names:
                if (m.op == "=" and m.value == "") or \
                   (m.op == "=~" and _re.fullmatch(m.value, "")) or \
                   (m.op == "!=" and m.value != "") or \
                   (m.op == "!~" and not _re.fullmatch(m.value, "")):
                    continue
                return np.zeros(0, dtype=np.int64)
            codes_arr, values = region.series.tag_codes(m.name)
            nv = len(values)
            vmask = np.zeros(nv + 1, dtype=bool)  # +1 slot for absent (-1)
            if m.op == "=":
                # exact: dictionary lookup instead of scanning values
                inv = region.series.inverted.get(m.name, {})
                got = np.zeros(n, dtype=bool)
                lst = inv.get(m.value)
                if lst:
                    got[np.asarray(lst, dtype=np.int64)] = True
            else:
                if m.op in ("=~", "!~"):
                    pat = _re.compile(m.value)
                    for i, v in enumerate(values):
                        vmask[i] = bool(pat.fullmatch(v))
                    vmask[nv] = bool(pat.fullmatch(""))  # absent label = ""
                else:  # !=
                    for i, v in enumerate(values):
                        vmask[i] = v != m.value
                    vmask[nv] = m.value != ""
                sel_ids = codes_arr.astype(np.int64)
                sel_ids[sel_ids < 0] = nv
                got = vmask[sel_ids]
                if m.op == "!~":
                    got = ~got
            mask = got if mask is None else (mask & got)
        if mask is None:
            return None
        return np.flatnonzero(mask)

    def _eval_subquery_range(self, sub: ast.Subquery, func: str, t0, step, T,
                             grid, param) -> PromMatrix:
        """expr[range:res] — evaluate the inner expression on its own
        res-aligned grid (Prometheus SubqueryExpr: start aligned UP to a
        multiple of the resolution), then treat the resulting matrix cells
        as samples for the enclosing range function. The window evaluation
        reuses the prom_range_eval kernel: NaN cells are compacted out into
        per-series ragged segments on device."""
        range_ms = int(sub.range_s * 1000)
        offset_ms = int(sub.offset_s * 1000)
        res = int(sub.step_s * 1000) if sub.step_s else step  # default: query step
        res = max(res, 1)
        inner_start = t0 - offset_ms - range_ms
        if inner_start % res:
            inner_start += res - inner_start % res
        inner_end = t0 + (T - 1) * step - offset_ms
        T2 = int((inner_end - inner_start) // res) + 1
        device = self.engine.config.device
        if T2 <= 0:
            return PromMatrix([], torch.zeros((0, T), dtype=torch.float64,
                                              device=device), grid)
        igrid = inner_start + np.arange(T2, dtype=np.int64) * res
        m = self._eval(sub.expr, inner_start, res, T2, igrid)
        if isinstance(m, PromScalar):
            m = PromMatrix([{}], _scalar_plane(m, T2, device), igrid)
        v = m.values
        S = int(v.shape[0])
        if S == 0:
            return PromMatrix([], torch.zeros((0, T), dtype=torch.float64,
                                              device=device), grid)
        mask = ~torch.isnan(v)
        counts = mask.sum(dim=1)
        seg_hi = torch.cumsum(counts, 0)
        seg_lo = seg_hi - counts
        ts_row = torch.as_tensor(igrid).to(v.device)
        ts_flat = ts_row.expand(S, T2)[mask]
        vals_flat = v[mask]
        mode = PROM_MODES[func]
        out = prom_range_eval(ts_flat.contiguous(), vals_flat.contiguous(),
                              seg_lo.contiguous(), seg_hi.contiguous(),
                              T, t0, step, range_ms, offset_ms, param, mode)
        return PromMatrix(m._labels, out, grid, label_cols=m.label_cols,
                          n_series=S)

    def _eval_selector(self, sel: ast.Selector, func: str, t0, step, T, grid,
                       range_ms, param) -> PromMatrix:
        if sel.at_s is not None:
            # @ modifier: evaluate once at the pinned time, broadcast over
            # the grid (Prometheus AtModifier semantics)
            at_ms = (t0 if sel.at_s == "start" else
                     t0 + (T - 1) * step if sel.at_s == "end" else
                     int(float(sel.at_s) * 1000))
            pinned = ast.Selector(sel.metric, sel.matchers, sel.range_s,
                                  sel.offset_s)
            m = self._eval_selector(pinned, func, at_ms, max(step, 1), 1,
                                    np.array([at_ms], dtype=np.int64),
                                    range_ms, param)
            vals = m.values if T == 1 else m.values.expand(-1, T).contiguous()
            return PromMatrix(m._labels, vals, grid,
                              label_cols=m.label_cols, n_series=m.S)
        st, field = self._resolve_table(sel)
        device = self.engine.config.device
        if st is None:
            return PromMatrix([], torch.zeros((0, T), dtype=torch.float64,
                                              device=device), grid)
        offset_ms = int(sel.offset_s * 1000)
        lo = t0 - offset_ms - range_ms - self.lookback_ms
        hi = t0 + (T - 1) * step - offset_ms + 1

        S_total = 0
        col_parts: dict[str, list] = {}   # tag -> [(codes slice, values)...] per region
        region_sizes: list[int] = []
        all_tags: list[str] = []
        parts = []          # per-region (ts, slots) sorted chunks
        vparts = []
        seg_counts = []
        for region in st.regions:
            codes = self._match_codes(region, sel)
            n_codes = len(region.series) if codes is None else len(codes)
            if n_codes == 0:
                continue
            base = S_total
            lut = np.full(len(region.series), -1, dtype=np.int32)
            if codes is None:
                lut[:] = base + np.arange(len(region.series), dtype=np.int32)
            else:
                ca = np.asarray(codes, dtype=np.int64)
                lut[ca] = base + np.arange(len(ca), dtype=np.int32)
            # factorized label columns for the matched codes (vectorized)
            sel_idx = np.arange(len(region.series)) if codes is None else \
                np.asarray(codes, dtype=np.int64)
            for t in region.series.tag_names:
                tcodes, tvalues = region.series.tag_codes(t)
                if t not in col_parts:
                    col_parts[t] = []
                    all_tags.append(t)
                col_parts[t].append((len(region_sizes), tcodes[sel_idx], tvalues))
            region_sizes.append(n_codes)
            S_total += n_codes
            lut_t = torch.as_tensor(lut, device=device)
            chunks = []
            all_sorted = True
            for src in region.scan_sources(lo, hi):
                p = src.field_pos.get(field)
                if p is None:
                    continue
                from greptimedb_amd.ops import filter_series_time
                mask = filter_series_time(src.ts, src.series,
                                          lut_t if codes is not None else None, lo, hi)
                idx = mask.nonzero(as_tuple=True)[0]
                if idx.numel() == 0:
                    continue
                chunks.append((src.ts[idx], src.series[idx], src.fields[p][idx]))
                all_sorted = all_sorted and src.sorted
            if not chunks:
                seg_counts.append((base, n_codes, 0))
                continue
            ts_t = torch.cat([c[0] for c in chunks])
            se_t = torch.cat([c[1] for c in chunks])
            v_t = torch.cat([c[2] for c in chunks])
            slots = lut_t[se_t.long()]
            ok = slots >= 0
            if not bool(ok.all()):
                ts_t, slots, v_t = ts_t[ok], slots[ok], v_t[ok]
            # (slot, ts) order required by the window kernel. The slot LUT is
            # monotone in code, so a single (series, ts)-sorted source is
            # already slot-sorted — skip the sort (the common SST-cache case).
            if not (len(chunks) == 1 and all_sorted):
                o = torch.argsort(ts_t, stable=True)
                perm = o[torch.argsort(slots[o], stable=True)]
                ts_t, slots, v_t = ts_t[perm], slots[perm], v_t[perm]
            keep = dedup_mark_last(slots.int().contiguous(), ts_t.contiguous())
            kidx = keep.nonzero(as_tuple=True)[0]
            if kidx.numel() != ts_t.numel():
                ts_t, slots, v_t = ts_t[kidx], slots[kidx], v_t[kidx]
            parts.append((ts_t, slots))
            vparts.append(v_t)
            seg_counts.append((base, n_codes, ts_t.numel()))

        S = S_total
        if S == 0:
            return PromMatrix([], torch.zeros((0, T), dtype=torch.float64,
                                              device=device), grid)
        # merge per-region factor columns (remap value ids into a shared
        # vocabulary). pd.factorize hash-codes the concatenated vocabs in C
        # (the python setdefault loop here cost ~1.3s at 4M series); the
        # merged columns cache on the engine until the series count moves.
        cache = getattr(self.engine, "_prom_label_cache", None)
        if cache is None:
            cache = self.engine._prom_label_cache = {}
        label_cols: dict = {}
        for t in all_tags:
            ck = (st.schema.name, t, S)
            hit = cache.get(ck)
            if hit is not None:
                label_cols[t] = hit
                continue
            import pandas as pd
            by_region = {ri: (c, v) for ri, c, v in col_parts[t]}
            vocab_arrays = []
            for ri, sz in enumerate(region_sizes):
                got = by_region.get(ri)
                vocab_arrays.append(
                    np.asarray(got[1], dtype=object) if got is not None
                    else np.empty(0, dtype=object))
            concat = np.concatenate(vocab_arrays) if vocab_arrays else                 np.empty(0, dtype=object)
            codes_flat, uniq = pd.factorize(concat)
            codes_flat = codes_flat.astype(np.int32)
            out_codes = np.full(S, -1, dtype=np.int32)
            off = 0
            voff = 0
            for ri, sz in enumerate(region_sizes):
                got = by_region.get(ri)
                if got is not None:
                    c, vals = got
                    remap = codes_flat[voff:voff + len(vals)]
                    voff += len(vals)
                    cc = c.astype(np.int64)
                    res = np.full(sz, -1, dtype=np.int32)
                    has = cc >= 0
                    res[has] = remap[cc[has]]
                    out_codes[off:off + sz] = res
                off += sz
            label_cols[t] = (out_codes, list(uniq))
            cache[ck] = label_cols[t]
            if len(cache) > 64:
                cache.pop(next(iter(cache)))
        if "__name__" not in label_cols:
            label_cols["__name__"] = (np.zeros(S, dtype=np.int32),
                                      [sel.metric or st.schema.name])
        ts_all = torch.cat([p[0] for p in parts]) if parts else \
            torch.zeros(0, dtype=torch.int64, device=device)
        slots_all = torch.cat([p[1] for p in parts]).long() if parts else \
            torch.zeros(0, dtype=torch.int64, device=device)
        vals_all = torch.cat(vparts) if vparts else \
            torch.zeros(0, dtype=torch.float64, device=device)
        # segments per slot (slots_all ascending because regions own
        # consecutive slot ranges and each part is slot-sorted)
        counts = torch.bincount(slots_all, minlength=S)
        seg_hi = torch.cumsum(counts, 0)
        seg_lo = seg_hi - counts
        mode = PROM_MODES["instant"] if func == "instant" else PROM_MODES[func]
        rng = self.lookback_ms if func == "instant" else range_ms
        out = prom_range_eval(ts_all.contiguous(), vals_all.contiguous(),
                              seg_lo.contiguous(), seg_hi.contiguous(),
                              T, t0, step, rng, offset_ms, param, mode)
        return PromMatrix(None, out, grid, label_cols=label_cols, n_series=S)

    # ------------------------------------------------------------ aggregate

    def _eval_aggregate(self, e: ast.Aggregate, t0, step, T, grid) -> PromMatrix:
        m = self._eval(e.expr, t0, step, T, grid)
        if isinstance(m, PromScalar):
            raise PlanQuery("aggregate over scalar")
        op = e.op
        if op in ("topk", "bottomk"):
            return self._topk(e, m, grid)
        # vectorized grouping over factorized label columns (no per-series
        # Python loop — required for 10M-series metric-engine queries)
        S = m.S
        cols = m.get_label_cols()
        if e.by is not None:
            gnames = list(e.by)
        elif e.without is not None:
            drop = set(e.without) | {"__name__"}
            gnames = [n for n in cols if n not in drop]
        else:
            gnames = []
        if not gnames or S == 0:
            gidx_arr = np.zeros(S, dtype=np.int64)
            out_labels = [{}]
        else:
            key = np.zeros(S, dtype=np.int64)
            mult = 1
            for n in gnames:
                codes, values = cols.get(n, (np.full(S, -1, dtype=np.int32), []))
                card = len(values) + 1
                if mult > (1 << 62) // max(card, 1):
                    _, key = np.unique(key, return_inverse=True)
                    mult = int(key.max()) + 1 if len(key) else 1
                key = key + (codes.astype(np.int64) + 1) * mult
                mult *= card
            uniq, first_idx, gidx_arr = np.unique(key, return_index=True,
                                                  return_inverse=True)
            out_labels = []
            for idx in first_idx:
                l = {}
                for n in gnames:
                    got = cols.get(n)
                    if got is None:
                        continue
                    c = got[0][idx]
                    if c >= 0:
                        l[n] = got[1][c]
                out_labels.append(l)
        G = max(len(out_labels), 1)
        dev = m.values.device
        gi = torch.as_tensor(gidx_arr, dtype=torch.int64, device=dev)
        gidx = gidx_arr  # for the quantile path below
        single_group = G == 1
        v = m.values
        present = ~torch.isnan(v)
        v0 = torch.where(present, v, torch.zeros_like(v))
        cnt = torch.zeros((G, T), dtype=torch.float64, device=dev)
        if m.S:
            if single_group:
                # index_add with one destination cell is a full-serialization
                # atomic collision on GPU (21s at 1M series) — reduce instead
                cnt[0] = present.double().sum(dim=0)
            else:
                cnt.index_add_(0, gi, present.double())

        if op == "count_values":
            # group additionally by the sample VALUE at each step; output
            # series = (group labels + {param: value}) (Prometheus semantics)
            label_name = e.param.value if e.param is not None else "value"
            vals_h = v.cpu().numpy()
            out_rows: dict = {}   # (g, value_repr) -> row index
            cols_out: list = []
            for t in range(T):
                col = vals_h[:, t]
                ok = ~np.isnan(col)
                if not ok.any():
                    continue
                pairs = np.stack([gidx[ok].astype(np.float64), col[ok]])
                uniq_p, counts_p = np.unique(pairs, axis=1, return_counts=True)
                for (g, val), c in zip(uniq_p.T, counts_p):
                    key = (int(g), float(val))
                    row = out_rows.setdefault(key, len(out_rows))
                    cols_out.append((row, t, float(c)))
            S2 = max(len(out_rows), 0)
            outm = torch.full((S2, T), float("nan"), dtype=torch.float64)
            for row, t, c in cols_out:
                outm[row, t] = c
            labels2 = []
            for (g, val) in out_rows:
                l = dict(out_labels[g]) if out_labels else {}
                l[str(label_name)] = f"{val:g}"
                labels2.append(l)
            if self.dist is not None:
                # gather per-rank count matrices, sum rows with equal
                # labels (counts are partials of disjoint series shards)
                keys = [tuple(sorted(l.items())) for l in labels2]
                keys_all, mat = self.dist.gather_matrix(keys, outm)
                uniq = sorted(set(keys_all))
                kmap = {k: i for i, k in enumerate(uniq)}
                m2 = torch.zeros((len(uniq), T), dtype=torch.float64)
                seen = torch.zeros((len(uniq), T), dtype=torch.bool)
                mat = torch.nan_to_num(mat.double(), nan=0.0)
                has = mat > 0
                for i, k in enumerate(keys_all):
                    m2[kmap[k]] += mat[i]
                    seen[kmap[k]] |= has[i]
                m2 = torch.where(seen, m2, torch.full_like(m2, float("nan")))
                labels2 = [dict(k) for k in uniq]
                outm = m2
            return PromMatrix(labels2, outm.to(dev), grid)

        if op == "quantile":
            q = e.param.value if isinstance(e.param, ast.NumberLit) else 0.5
            if self.dist is not None:
                # quantile is not mergeable from partials: gather member
                # series values per group (reference ships all series to
                # the frontend for the final quantile the same way)
                keys = [tuple(sorted(out_labels[int(g)].items()))
                        for g in gidx] if S else []
                keys_all, v_all = self.dist.gather_matrix(keys, v)
                uniq = sorted(set(keys_all))
                kmap = {k: i for i, k in enumerate(uniq)}
                gidx = np.array([kmap[k] for k in keys_all], dtype=np.int64)
                v = v_all.double().to(dev)
                out_labels = [dict(k) for k in uniq]
                G = max(len(out_labels), 1)
                cnt = torch.zeros((G, T), dtype=torch.float64, device=dev)
                if len(gidx):
                    cnt.index_add_(0, torch.as_tensor(gidx, device=dev),
                                   (~torch.isnan(v)).double())
            out = torch.full((G, T), float("nan"), dtype=torch.float64, device=dev)
            for g in range(G):
                rows = [i for i, x in enumerate(gidx) if x == g]
                out[g] = torch.nanquantile(v[rows].float(), q, dim=0).double()
            out = torch.where(cnt > 0, out, torch.full_like(out, float("nan")))
            return PromMatrix(out_labels, out, grid)

        # partial planes (mergeable across ranks)
        s = torch.zeros((G, T), dtype=torch.float64, device=dev)
        sq = torch.zeros((G, T), dtype=torch.float64, device=dev) \
            if op in ("stddev", "stdvar") else None
        mn = mx = None
        if m.S:
            if single_group:
                s[0] = v0.sum(dim=0)
                if sq is not None:
                    sq[0] = (v0 * v0).sum(dim=0)
            else:
                s.index_add_(0, gi, v0)
                if sq is not None:
                    sq.index_add_(0, gi, v0 * v0)
        if op in ("min", "max"):
            fill = float("inf") if op == "min" else float("-inf")
            t = torch.full((G, T), fill, dtype=torch.float64, device=dev)
            if m.S:
                vm = torch.where(present, v, torch.full_like(v, fill))
                if single_group:
                    t[0] = vm.amin(dim=0) if op == "min" else vm.amax(dim=0)
                else:
                    t.index_reduce_(0, gi, vm, "amin" if op == "min" else "amax",
                                    include_self=True)
            if op == "min":
                mn = t
            else:
                mx = t

        if self.dist is not None:
            out_labels, cnt, s, sq, mn, mx = self.dist.merge_prom_planes(
                [tuple(sorted(l.items())) for l in out_labels], cnt, s, sq, mn, mx)
            out_labels = [dict(k) for k in out_labels]
            G = max(len(out_labels), 1)

        nan = float("nan")
        if op == "count":
            out = torch.where(cnt > 0, cnt, torch.full_like(cnt, nan))
        elif op == "sum":
            out = torch.where(cnt > 0, s, torch.full_like(s, nan))
        elif op == "avg":
            out = torch.where(cnt > 0, s / cnt, torch.full_like(s, nan))
        elif op in ("stddev", "stdvar"):
            mean = s / cnt.clamp_min(1)
            var = (sq / cnt.clamp_min(1) - mean * mean).clamp_min(0)
            out = torch.where(cnt > 0, var if op == "stdvar" else var.sqrt(),
                              torch.full_like(var, nan))
        elif op == "group":
            out = torch.where(cnt > 0, torch.ones_like(cnt), torch.full_like(cnt, nan))
        elif op == "min":
            out = torch.where(cnt > 0, mn, torch.full_like(mn, nan))
        elif op == "max":
            out = torch.where(cnt > 0, mx, torch.full_like(mx, nan))
        else:
            raise PlanQuery(f"promql: unsupported aggregation {op}")
        return PromMatrix(out_labels, out, grid)

    def _topk(self, e: ast.Aggregate, m: PromMatrix, grid) -> PromMatrix:
        k = int(e.param.value) if isinstance(e.param, ast.NumberLit) else 1
        v = m.values
        if m.S == 0 or k <= 0:
            return m
        desc = e.op == "topk"
        # rank per timestep; keep sample only when within top/bottom k
        filled = torch.where(torch.isnan(v),
                             torch.full_like(v, float("-inf") if desc else float("inf")), v)
        order = torch.argsort(filled, dim=0, descending=desc)
        rank = torch.empty_like(order)
        ar = torch.arange(m.S, device=v.device)[:, None].expand_as(order)
        rank.scatter_(0, order, ar)
        keep = rank < k
        out = torch.where(keep & ~torch.isnan(v), v, torch.full_like(v, float("nan")))
        used = (~torch.isnan(out)).any(dim=1)
        idx = used.nonzero(as_tuple=True)[0]
        return PromMatrix([m.labels[int(i)] for i in idx], out[idx], grid)

    # ------------------------------------------------------------ binop

    def _eval_binop(self, e: ast.BinOp, t0, step, T, grid):
        l = self._eval(e.left, t0, step, T, grid)
        r = self._eval(e.right, t0, step, T, grid)
        if isinstance(l, PromScalar) and isinstance(r, PromScalar):
            if isinstance(l.value, torch.Tensor) or \
                    isinstance(r.value, torch.Tensor):
                a = _scalar_plane(l, T)
                b = _scalar_plane(r, T).to(a.device)
                res, _keep = _vector_op(e.op, a, b, True)
                return PromScalar(res[0], grid)
            return PromScalar(_scalar_op(e.op, l.value, r.value), grid)
        if isinstance(l, PromScalar) or isinstance(r, PromScalar):
            mat, sc, flipped = (r, l, True) if isinstance(l, PromScalar) else (l, r, False)
            a = mat.values
            b = _scalar_plane(sc, T, a.device).expand_as(a) \
                if isinstance(sc.value, torch.Tensor) else \
                torch.as_tensor(float(sc.value), dtype=torch.float64, device=a.device)
            if flipped:
                res, keep = _vector_op(e.op, b.expand_as(a), a, e.bool_modifier)
            else:
                res, keep = _vector_op(e.op, a, b, e.bool_modifier)
            if e.op in _CMP_OPS and not e.bool_modifier:
                res = torch.where(keep, mat.values, torch.full_like(res, float("nan")))
            return _matrix_map(mat, res, drop_name=True)
        # vector-vector: one-to-one on matching label sets
        if e.op in ("and", "or", "unless"):
            return self._set_op(e.op, l, r, grid)
        if e.group_left or e.group_right:
            # many-to-one: the "one" side must be unique per match key; every
            # "many" row keeps its full labels (Prometheus group_left/right)
            many, one = (l, r) if e.group_left else (r, l)
            one_map: dict = {}
            for i, x in enumerate(one.labels):
                k = _match_key(x, e.on, e.ignoring)
                if k in one_map:
                    raise PlanQuery('promql: duplicate series on the "one" '
                                    'side of group_left/group_right')
                one_map[k] = i
            rows_many, rows_one = [], []
            for i, x in enumerate(many.labels):
                j = one_map.get(_match_key(x, e.on, e.ignoring))
                if j is not None:
                    rows_many.append(i)
                    rows_one.append(j)
            if not rows_many:
                return PromMatrix([], torch.zeros(
                    (0, T), dtype=torch.float64, device=l.values.device), grid)
            mi = torch.as_tensor(rows_many, device=l.values.device)
            oi = torch.as_tensor(rows_one, device=l.values.device)
            a_m, b_o = many.values[mi], one.values[oi]
            a, b = (a_m, b_o) if e.group_left else (b_o, a_m)
            res, keep = _vector_op(e.op, a, b, e.bool_modifier)
            if e.op in _CMP_OPS and not e.bool_modifier:
                res = torch.where(keep, a, torch.full_like(res, float("nan")))
            labels = [_drop_name(many.labels[int(i)]) for i in mi]
            return PromMatrix(labels, res, grid)
        lk, rk = {}, {}
        for side, key_map in ((l, lk), (r, rk)):
            for i, x in enumerate(side.labels):
                k = _match_key(x, e.on, e.ignoring)
                if k in key_map:
                    raise PlanQuery("promql: many-to-one matching needs "
                                    "group_left/group_right")
                key_map[k] = i
        common = [k for k in lk if k in rk]
        if not common:
            return PromMatrix([], torch.zeros((0, T), dtype=torch.float64,
                                              device=l.values.device), grid)
        li = torch.as_tensor([lk[k] for k in common], device=l.values.device)
        ri = torch.as_tensor([rk[k] for k in common], device=l.values.device)
        a = l.values[li]
        b = r.values[ri]
        res, keep = _vector_op(e.op, a, b, e.bool_modifier)
        if e.op in _CMP_OPS and not e.bool_modifier:
            res = torch.where(keep, a, torch.full_like(res, float("nan")))
        labels = [_drop_name(l.labels[int(i)]) for i in li]
        return PromMatrix(labels, res, grid)

    def _set_op(self, op, l: PromMatrix, r: PromMatrix, grid):
        rk = {_match_key(x, None, None) for x in r.labels}
        if op == "and":
            idx = [i for i, x in enumerate(l.labels) if _match_key(x, None, None) in rk]
            rpresent = torch.zeros_like(l.values[0], dtype=torch.bool) if r.S == 0 else None
            out_rows = []
            for i in idx:
                out_rows.append(l.values[i])
            vals = torch.stack(out_rows) if out_rows else \
                torch.zeros((0, l.values.shape[1]), dtype=torch.float64,
                            device=l.values.device)
            return PromMatrix([l.labels[i] for i in idx], vals, grid)
        if op == "unless":
            idx = [i for i, x in enumerate(l.labels) if _match_key(x, None, None) not in rk]
            vals = l.values[idx] if idx else torch.zeros(
                (0, l.values.shape[1]), dtype=torch.float64, device=l.values.device)
            return PromMatrix([l.labels[i] for i in idx], vals, grid)
        # or: left series + right series not in left
        lkeys = {_match_key(x, None, None) for x in l.labels}
        extra = [i for i, x in enumerate(r.labels) if _match_key(x, None, None) not in lkeys]
        labels = list(l.labels) + [r.labels[i] for i in extra]
        vals = torch.cat([l.values, r.values[extra]]) if extra else l.values
        return PromMatrix(labels, vals, grid)


_CMP_OPS = {"==", "!=", "<", "<=", ">", ">="}


def _scalar_op(op, a, b):
    if op == "+":
        return a + b
    if op == "-":
        return a - b
    if op == "*":
        return a * b
    if op == "/":
        return a / b if b != 0 else math.inf if a > 0 else -math.inf if a < 0 else math.nan
    if op == "%":
        return math.fmod(a, b) if b != 0 else math.nan
    if op == "^":
        return a ** b
    if op == "==":
        return 1.0 if a == b else 0.0
    if op == "!=":
        return 1.0 if a != b else 0.0
    if op == "<":
        return 1.0 if a < b else 0.0
    if op == "<=":
        return 1.0 if a <= b else 0.0
    if op == ">":
        return 1.0 if a > b else 0.0
    if op == ">=":
        return 1.0 if a >= b else 0.0
    raise PlanQuery(f"promql: op {op}")


def _vector_op(op, a, b, bool_mod):
    if op == "+":
        return a + b, None
    if op == "-":
        return a - b, None
    if op == "*":
        return a * b, None
    if op == "/":
        return a / b, None
    if op == "%":
        return torch.fmod(a, b), None
    if op == "^":
        return a ** b, None
    if op in _CMP_OPS:
        keep = {"==": a == b, "!=": a != b, "<": a < b,
                "<=": a <= b, ">": a > b, ">=": a >= b}[op]
        keep &= ~torch.isnan(a)
        if torch.is_tensor(b):
            keep &= ~torch.isnan(b)
        if bool_mod:
            nanmask = torch.isnan(a)
            res = keep.double()
            res = torch.where(nanmask, torch.full_like(res, float("nan")), res)
            return res, keep
        return a, keep
    raise PlanQuery(f"promql: op {op}")



def _matrix_map(m: "PromMatrix", values=None, drop_name=False) -> "PromMatrix":
    """New matrix with transformed values, keeping labels columnar/lazy."""
    v = m.values if values is None else values
    if m.label_cols is not None:
        cols = m.label_cols
        if drop_name:
            cols = {k: c for k, c in cols.items() if k != "__name__"}
        return PromMatrix(None, v, m.grid, label_cols=cols, n_series=m.S)
    labels = [_drop_name(l) for l in m.labels] if drop_name else m.labels
    return PromMatrix(labels, v, m.grid)


def _drop_name(l: dict) -> dict:
    return {k: v for k, v in l.items() if k != "__name__"}


def _group_labels(l: dict, by, without) -> dict:
    if by is not None:
        return {k: l[k] for k in by if k in l}
    if without is None:
        return {}  # plain sum(...) collapses all labels
    drop = set(without) | {"__name__"}
    return {k: v for k, v in l.items() if k not in drop}


def _match_key(l: dict, on, ignoring):
    if on is not None:
        return tuple(sorted((k, v) for k, v in l.items() if k in on))
    drop = set(ignoring or []) | {"__name__"}
    return tuple(sorted((k, v) for k, v in l.items() if k not in drop))
