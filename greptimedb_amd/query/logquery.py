"""Log query DSL → SQL translation (HTTP /v1/logs).

Reference parity: src/log-query/src/log_query.rs — LogQuery {table,
time_filter {start/end/span}, limit {skip/fetch}, columns, filters
[ColumnFilters {expr, filters: [ContentFilter]}], exprs} executed by the
query engine (src/query/src/log_query/). Here the DSL compiles to the SQL
surface (MATCHES for fulltext terms, LIKE for prefix/postfix/contains),
which runs on the same GPU scan path as every other query.
"""

from __future__ import annotations

from greptimedb_amd.utils.errors import InvalidArguments
from greptimedb_amd.utils.timeutil import parse_ts_ms


def _q(s: str) -> str:
    return "'" + str(s).replace("'", "''") + "'"


def _like_escape(s: str) -> str:
    return str(s).replace("%", r"\%").replace("_", r"\_")


def _content_filter_sql(col: str, f: dict | str) -> str:
    """One ContentFilter (log_query.rs:383) → SQL predicate."""
    if isinstance(f, str):      # bare string = Contains
        f = {"Contains": f}
    (kind, val), = f.items()
    kind_l = kind.lower()
    if kind_l == "exact":
        # exact term match → fulltext MATCHES (tokenized term)
        return f"matches({col}, {_q(val)})"
    if kind_l == "contains":
        return f"{col} LIKE {_q('%' + _like_escape(val) + '%')}"
    if kind_l == "prefix":
        return f"{col} LIKE {_q(_like_escape(val) + '%')}"
    if kind_l == "postfix":
        return f"{col} LIKE {_q('%' + _like_escape(val))}"
    if kind_l == "equal":
        if isinstance(val, dict):
            (_t, v), = val.items()
            val = v
        if isinstance(val, bool):
            return f"{col} = {str(val).upper()}"
        if isinstance(val, (int, float)):
            return f"{col} = {val}"
        return f"{col} = {_q(val)}"
    if kind_l in ("greatthan", "greaterthan", "gt"):
        v, inc = _value_inclusive(val)
        return f"{col} {'>=' if inc else '>'} {v}"
    if kind_l in ("lessthan", "lt"):
        v, inc = _value_inclusive(val)
        return f"{col} {'<=' if inc else '<'} {v}"
    if kind_l == "in":
        vals = ", ".join(_q(v) if isinstance(v, str) else str(v) for v in val)
        return f"{col} IN ({vals})"
    if kind_l == "exist":
        return f"{col} IS NOT NULL"
    if kind_l == "compound":
        parts, op = val if isinstance(val, list) else (val.get("filters"),
                                                      val.get("op", "and"))
        if isinstance(val, dict):
            parts = val.get("filters", [])
            op = str(val.get("op", "and")).lower()
        else:
            op = "and"
        sub = [_content_filter_sql(col, p) for p in parts]
        return "(" + f" {op.upper()} ".join(sub) + ")"
    raise InvalidArguments(f"unsupported content filter {kind}")


def _value_inclusive(val):
    if isinstance(val, dict):
        inc = bool(val.get("inclusive", False))
        v = val.get("value")
    else:
        inc, v = False, val
    if isinstance(v, str):
        return _q(v), inc
    return v, inc


def _expr_col(e) -> str:
    """LogExpr → column name (NamedIdent or {NamedIdent: name})."""
    if isinstance(e, str):
        return e
    if isinstance(e, dict):
        if "NamedIdent" in e:
            return str(e["NamedIdent"])
        if "column_name" in e:
            return str(e["column_name"])
    raise InvalidArguments(f"unsupported log expr {e}")


def logquery_to_sql(q: dict) -> str:
    table = q.get("table")
    if isinstance(table, dict):
        table = table.get("table_name") or table.get("table")
    if not table:
        raise InvalidArguments("log query needs a table")
    tf = q.get("time_filter") or {}
    conds = []
    ts_col = q.get("time_column", "ts")
    start, end = tf.get("start"), tf.get("end")
    if start is not None:
        conds.append(f"{ts_col} >= {int(parse_ts_ms(start))}")
    if end is not None:
        conds.append(f"{ts_col} < {int(parse_ts_ms(end))}")
    elif start is not None and tf.get("span"):
        from greptimedb_amd.query.parser import parse_interval_text
        span = parse_interval_text(str(tf["span"]))
        conds.append(f"{ts_col} < {int(parse_ts_ms(start)) + span}")
    for cf in q.get("filters") or []:
        col = _expr_col(cf.get("expr") or cf.get("column_name"))
        fl = cf.get("filters") or []
        for f in fl:
            conds.append(_content_filter_sql(col, f))
    cols = q.get("columns") or []
    proj = ", ".join(_expr_col(c) for c in cols) if cols else "*"
    sql = f"SELECT {proj} FROM {table}"
    if conds:
        sql += " WHERE " + " AND ".join(conds)
    sql += f" ORDER BY {ts_col}"
    lim = q.get("limit") or {}
    fetch = lim.get("fetch", 1000)
    skip = lim.get("skip", 0)
    sql += f" LIMIT {int(fetch)}"
    if skip:
        sql += f" OFFSET {int(skip)}"
    return sql
