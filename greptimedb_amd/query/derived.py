"""Derived-table evaluation: SELECT over a materialized QueryResult.

Reference parity: DataFusion executes CTEs, views and set operations over
intermediate record batches (src/query planner); here the same surface runs
over materialized host columns — the frontend-side half of the MergeScan
split. Used for:
  * WITH ctes (`Select.ctes`)
  * views (CREATE VIEW — engine.views, reference src/common/meta ddl
    create_view + src/catalog view support)
  * derived tables: FROM (SELECT ...) alias
  * UNION / UNION ALL / EXCEPT / INTERSECT (reference setops cases)
"""

from __future__ import annotations

import numpy as np

from greptimedb_amd.query import ast
from greptimedb_amd.utils.errors import PlanQuery

AGGS = {"count", "sum", "min", "max", "avg", "mean", "last_value",
        "first_value", "stddev", "var",
        # sketch/approx aggregates (reference aggrs/approximate)
        "hll", "hll_merge", "uddsketch_state", "uddsketch_merge",
        "approx_percentile", "median"}

SKETCH_SCALARS = {"hll_count", "uddsketch_calc"}


def _contains_agg(e) -> bool:
    if isinstance(e, ast.Func):
        if e.name.lower() in AGGS:
            return True
        return any(_contains_agg(a) for a in e.args)
    if isinstance(e, (ast.Lit, ast.Col, ast.Star)):
        return False
    if isinstance(e, ast.BinOp):
        return _contains_agg(e.left) or _contains_agg(e.right)
    if isinstance(e, ast.UnaryOp):
        return _contains_agg(e.operand)
    return False


def _as_cols(res) -> tuple[dict, list, int]:
    """QueryResult → ({name: np array}, kinds, n)."""
    cols = {}
    n = len(res.columns[0]) if res.columns else 0
    for name, col, kind in zip(res.names, res.columns, res.kinds):
        a = np.asarray(col)
        if a.dtype.kind in "iuf" and kind != "ts":
            a = a.astype(np.float64)
        elif a.dtype.kind not in "iuf":
            a = np.asarray(col, dtype=object)
        cols[name] = a
    return cols, list(res.kinds), n


def _agg_value(fname: str, vals: np.ndarray):
    fname = fname.lower()
    if fname == "count":
        if vals.dtype == object:
            return float(sum(1 for v in vals if v is not None))
        return float(np.count_nonzero(~np.isnan(vals.astype(np.float64))))
    v = vals.astype(np.float64) if vals.dtype != object else \
        np.array([x for x in vals if x is not None], dtype=np.float64)
    v = v[~np.isnan(v)]
    if len(v) == 0:
        return None
    return {"sum": np.sum, "min": np.min, "max": np.max, "avg": np.mean,
            "mean": np.mean, "stddev": np.std, "var": np.var,
            "last_value": lambda x: x[-1],
            "first_value": lambda x: x[0]}[fname](v)


def _eval_scalar(e, col_data: dict, rows: np.ndarray):
    """Evaluate an expression (possibly containing aggregates) over the
    row subset `rows` to a scalar (aggregate context)."""
    from greptimedb_amd.query import executor as X
    fn = e.name.lower() if isinstance(e, ast.Func) else None
    if fn in ("hll", "hll_merge", "uddsketch_state", "uddsketch_merge",
              "approx_percentile", "median"):
        from greptimedb_amd.query.sketches import Hll, UddSketch
        # value column: first arg for approx_percentile/median (col, p);
        # LAST arg for the sketch builders (params lead: b, e, col)
        arg = e.args[0] if fn in ("approx_percentile", "median") else e.args[-1]
        vals = np.atleast_1d(np.asarray(X._np_raw(arg, col_data)))[rows]
        if fn == "hll":
            return Hll().add_values(vals.tolist()).dumps()
        if fn == "hll_merge":
            h = Hll()
            for st_ in vals:
                if st_ is not None:
                    h.merge(Hll.loads(st_))
            return h.dumps()
        if fn == "uddsketch_state":
            nb = int(X._eval_const(e.args[0])) if len(e.args) > 2 else 128
            al = float(X._eval_const(e.args[1])) if len(e.args) > 2 else 0.01
            return UddSketch(nb, al).add_values(
                vals.astype(np.float64).tolist()).dumps()
        if fn == "uddsketch_merge":
            out = None
            for st_ in vals:
                if st_ is None:
                    continue
                sk = UddSketch.loads(st_)
                out = sk if out is None else out.merge(sk)
            return (out or UddSketch()).dumps()
        # approx_percentile(col, p) / median(col): exact selection (our
        # "approximation" is exact — strictly stronger than the sketch)
        p = float(X._eval_const(e.args[1])) if fn == "approx_percentile"             else 0.5
        fv = vals.astype(np.float64)
        fv = fv[~np.isnan(fv)]
        return float(np.quantile(fv, p)) if len(fv) else None
    if fn in SKETCH_SCALARS:
        from greptimedb_amd.query.sketches import hll_count, uddsketch_calc
        if fn == "hll_count":
            st_ = _eval_scalar(e.args[0], col_data, rows)
            return None if st_ is None else hll_count(st_)
        q = float(X._eval_const(e.args[0]))
        st_ = _eval_scalar(e.args[1], col_data, rows)
        return None if st_ is None else uddsketch_calc(q, st_)
    if isinstance(e, ast.Func) and e.name.lower() in AGGS:
        if e.name.lower() == "count" and (not e.args or
                                          isinstance(e.args[0], ast.Star)):
            return float(len(rows))
        arg = X._np_raw(e.args[0], col_data)
        vals = np.asarray(arg)[rows]
        if e.distinct:
            vals = np.array(sorted(set(vals.tolist()),
                                   key=lambda x: (x is None, x)), dtype=vals.dtype)
        return _agg_value(e.name, vals)
    if isinstance(e, ast.BinOp):
        l = _eval_scalar(e.left, col_data, rows)
        r = _eval_scalar(e.right, col_data, rows)
        if l is None or r is None:
            return None
        return X._np_binop(e.op, l, r)
    if isinstance(e, ast.UnaryOp) and e.op == "-":
        v = _eval_scalar(e.operand, col_data, rows)
        return None if v is None else -v
    if isinstance(e, ast.Lit):
        return e.value
    if isinstance(e, ast.Col):
        vals = col_data[e.name][rows]
        return vals[0] if len(vals) else None
    from greptimedb_amd.query import executor as X2
    sub = {k: v[rows] for k, v in col_data.items()}
    out = X2._np_raw(e, sub)
    return out[0] if np.ndim(out) == 1 and len(out) else out


def select_over_result(sel: ast.Select, base, base_kinds=None):
    """Run a Select (no joins) over a materialized QueryResult `base`."""
    from greptimedb_amd.query import executor as X
    col_data, kinds, n = _as_cols(base)
    kind_of = dict(zip(base.names, base.kinds))
    alias_map = {a: e for e, a in sel.projections if a}

    def resolve(e):
        """Aliases in GROUP BY/ORDER BY/HAVING refer to projection exprs."""
        if isinstance(e, ast.Col) and e.name in alias_map and \
                e.name not in col_data:
            return alias_map[e.name]
        return e

    rows = np.arange(n)
    if sel.where is not None:
        mask = X._eval_np_cond(sel.where, col_data)
        rows = rows[np.atleast_1d(mask)[: n]]

    # expand * projections
    projections = []
    for e, a in sel.projections:
        if isinstance(e, ast.Star):
            projections += [(ast.Col(nm), nm) for nm in base.names]
        else:
            projections.append((e, a))

    has_agg = any(_contains_agg(e) for e, _ in projections) or sel.group_by
    out_names, out_cols, out_kinds = [], [], []

    if has_agg:
        def resolve_g(g):
            g = resolve(g)
            # positional GROUP BY n → n-th projection
            if isinstance(g, ast.Lit) and isinstance(g.value, int) and \
                    1 <= g.value <= len(projections):
                return resolve(projections[g.value - 1][0])
            return g
        gexprs = [resolve_g(g) for g in sel.group_by]
        if gexprs:
            keys = [np.asarray(X._np_raw(g, col_data))[rows] for g in gexprs]
            groups: dict = {}
            for i in range(len(rows)):
                k = tuple(k_[i] for k_ in keys)
                groups.setdefault(k, []).append(i)
            items = sorted(groups.items(),
                           key=lambda kv: tuple((v is None, v) for v in kv[0]))
        else:
            items = [((), list(range(len(rows))))]
        if sel.having is not None:
            hv = resolve(sel.having)
            items = [(k, idxs) for k, idxs in items
                     if _truthy(_eval_scalar(_resolve_tree(hv, alias_map,
                                                           col_data),
                                             col_data, rows[idxs]))]
        for pi, (e, a) in enumerate(projections):
            name = a or _name_of(e, pi)
            vals = []
            for k, idxs in items:
                sub_rows = rows[np.asarray(idxs, dtype=np.int64)]
                if _contains_agg(e):
                    vals.append(_eval_scalar(e, col_data, sub_rows))
                else:
                    # group key expr: constant within the group
                    gv = np.asarray(X._np_raw(resolve(e), col_data))[sub_rows]
                    vals.append(gv[0] if len(gv) else None)
            out_names.append(name)
            out_cols.append(vals)
            out_kinds.append(kind_of.get(getattr(e, "name", None), ""))
    else:
        for pi, (e, a) in enumerate(projections):
            name = a or _name_of(e, pi)
            v = X._np_raw(e, {k: c[rows] for k, c in col_data.items()})
            if np.ndim(v) == 0:
                v = np.full(len(rows), v)
            out_names.append(name)
            out_cols.append(np.asarray(v))
            out_kinds.append(kind_of.get(getattr(e, "name", None), ""))

    res_cols = [np.asarray(c, dtype=object) if _is_objy(c) else
                np.asarray(c) for c in out_cols]
    m = len(res_cols[0]) if res_cols else 0

    # ORDER BY expressions that aren't output columns: evaluate them as
    # hidden columns aligned with the output rows (pre-distinct)
    name_idx = {nm: i for i, nm in enumerate(out_names)}
    order_extra: dict[int, np.ndarray] = {}
    if sel.order_by and m:
        for oi, (e, _desc) in enumerate(sel.order_by):
            e = resolve(e)
            if isinstance(e, ast.Col) and e.name in name_idx:
                continue
            if isinstance(e, ast.Lit) and isinstance(e.value, int):
                continue
            if has_agg:
                vals = []
                for k, idxs in items:
                    sub_rows = rows[np.asarray(idxs, dtype=np.int64)]
                    if _contains_agg(e):
                        vals.append(_eval_scalar(e, col_data, sub_rows))
                    else:
                        gv = np.asarray(X._np_raw(e, col_data))[sub_rows]
                        vals.append(gv[0] if len(gv) else None)
                order_extra[oi] = np.asarray(vals, dtype=object)
            else:
                v = X._np_raw(e, {k: c[rows] for k, c in col_data.items()})
                if np.ndim(v) == 0:
                    v = np.full(len(rows), v)
                order_extra[oi] = np.asarray(v)

    if sel.distinct and m:
        seen, keep = set(), []
        for i in range(m):
            k = tuple(_hashable(c[i]) for c in res_cols)
            if k not in seen:
                seen.add(k)
                keep.append(i)
        ka = np.asarray(keep, dtype=np.int64)
        res_cols = [c[ka] for c in res_cols]
        order_extra = {oi: c[ka] for oi, c in order_extra.items()}
        m = len(keep)

    if sel.order_by and m:
        idx = list(range(m))

        def key_for(i):
            ks = []
            for oi, (e, desc) in enumerate(sel.order_by):
                e = resolve(e)
                if oi in order_extra:
                    v = order_extra[oi][i]
                elif isinstance(e, ast.Col) and e.name in name_idx:
                    v = res_cols[name_idx[e.name]][i]
                elif isinstance(e, ast.Lit) and isinstance(e.value, int):
                    v = res_cols[e.value - 1][i]
                else:
                    raise PlanQuery("derived ORDER BY supports output columns")
                ks.append(_sort_key(v, desc))
            return tuple(ks)
        idx.sort(key=key_for)
        res_cols = [c[np.asarray(idx, dtype=np.int64)] for c in res_cols]

    off = sel.offset or 0
    if off:
        res_cols = [c[off:] for c in res_cols]
    if sel.limit is not None:
        res_cols = [c[: sel.limit] for c in res_cols]

    from greptimedb_amd.query.executor import QueryResult
    return QueryResult(out_names, [list(c) for c in res_cols], out_kinds)


class _Desc:
    __slots__ = ("v",)

    def __init__(self, v):
        self.v = v

    def __lt__(self, other):
        a, b = self.v, other.v
        if a is None:
            return False
        if b is None:
            return True
        return b < a


def _sort_key(v, desc):
    if isinstance(v, (np.floating, float)) and np.isnan(v):
        v = None
    key = (v is None, v)
    return _Desc(key) if desc else key


def _truthy(v) -> bool:
    return bool(v) and v is not None


def _hashable(v):
    return None if v is None else (float(v) if isinstance(v, (int, float,
                                                              np.number))
                                   else str(v))


def _is_objy(c) -> bool:
    try:
        a = np.asarray(c)
    except ValueError:
        return True
    return a.dtype == object or a.dtype.kind in "US" or \
        any(v is None for v in (c if isinstance(c, list) else a.tolist()))


def _resolve_tree(e, alias_map, col_data):
    if isinstance(e, ast.Col) and e.name in alias_map and e.name not in col_data:
        return alias_map[e.name]
    if isinstance(e, ast.BinOp):
        return ast.BinOp(e.op, _resolve_tree(e.left, alias_map, col_data),
                         _resolve_tree(e.right, alias_map, col_data))
    return e


def _name_of(e, i) -> str:
    if isinstance(e, ast.Col):
        return e.name
    if isinstance(e, ast.Func):
        inner = ", ".join(_name_of(a, 0) if not isinstance(a, ast.Star)
                          else "*" for a in e.args)
        if e.distinct:
            inner = "DISTINCT " + inner
        return f"{e.name}({inner})"
    if isinstance(e, ast.Lit):
        return str(e.value)
    if isinstance(e, ast.BinOp):
        return f"{_name_of(e.left, i)} {e.op} {_name_of(e.right, i)}"
    if isinstance(e, ast.UnaryOp):
        return f"{e.op}{_name_of(e.operand, i)}"
    if isinstance(e, ast.Case):
        return "case"
    return f"col{i}"


# ------------------------------------------------------------------ setops

def eval_setop(op: str, all_rows: bool, left, right):
    """UNION [ALL] / EXCEPT / INTERSECT over two QueryResults (positional
    column alignment, left side names win — SQL semantics)."""
    from greptimedb_amd.query.executor import QueryResult
    if len(left.names) != len(right.names):
        raise PlanQuery(f"{op.upper()} sides have different column counts")
    ncol = len(left.names)
    lrows = list(zip(*[list(c) for c in left.columns])) if left.columns else []
    rrows = list(zip(*[list(c) for c in right.columns])) if right.columns else []

    def key(row):
        return tuple(_hashable(v) for v in row)

    if op == "union":
        rows = lrows + rrows
        if not all_rows:
            seen, uniq = set(), []
            for r in rows:
                k = key(r)
                if k not in seen:
                    seen.add(k)
                    uniq.append(r)
            rows = uniq
    elif op == "except":
        rset = {key(r) for r in rrows}
        rows, seen = [], set()
        for r in lrows:
            k = key(r)
            if k not in rset and (all_rows or k not in seen):
                seen.add(k)
                rows.append(r)
    elif op == "intersect":
        rset = {key(r) for r in rrows}
        rows, seen = [], set()
        for r in lrows:
            k = key(r)
            if k in rset and (all_rows or k not in seen):
                seen.add(k)
                rows.append(r)
    else:
        raise PlanQuery(f"unknown set op {op}")
    cols = [list(c) for c in zip(*rows)] if rows else [[] for _ in range(ncol)]
    return QueryResult(list(left.names), cols, list(left.kinds))
