"""Multi-dimensional partition rules — `PARTITION ON COLUMNS (...)`.

Reference parity: src/partition/src/multi_dim.rs (MultiDimPartitionRule:
`find_region` for one row :102, vectorized `split_record_batch` :226) and
the PARTITION ON COLUMNS statement (src/sql CREATE TABLE). Each partition
is defined by a boolean expression over the partition columns; a row goes
to the first region whose expression matches. The rule is serialized into
the table schema options so it survives restarts and is re-resolved by
every Ingestor.

MI355X mapping (P1 axis): partition index → region; with
world>1 the owning rank is `partition % world` (write fan-out ships rows
to the owner — parallel/write_fanout.py).
"""

from __future__ import annotations

import json

import numpy as np

from greptimedb_amd.engine.series import tsid_hash


class PartitionExpr:
    """Boolean expr tree over partition columns.

    node = ("col", name) | ("lit", value) |
           (op, left, right) with op ∈ {lt le gt ge eq ne and or}
    """

    OPS = {"<": "lt", "<=": "le", ">": "gt", ">=": "ge", "=": "eq",
           "==": "eq", "!=": "ne", "<>": "ne", "and": "and", "or": "or"}
    _SQL = {"lt": "<", "le": "<=", "gt": ">", "ge": ">=", "eq": "=",
            "ne": "!=", "and": "AND", "or": "OR"}

    def __init__(self, node):
        self.node = node

    # ------------------------------------------------------------- build
    @staticmethod
    def from_ast(e) -> "PartitionExpr":
        """From the SQL parser's ast.Expr (BinOp/Col/Lit only)."""
        from greptimedb_amd.query import ast

        def conv(x):
            if isinstance(x, ast.Col):
                return ("col", x.name)
            if isinstance(x, ast.Lit):
                return ("lit", x.value)
            if isinstance(x, ast.BinOp):
                op = PartitionExpr.OPS.get(x.op.lower())
                if op is None:
                    raise ValueError(f"unsupported partition op {x.op!r}")
                return (op, conv(x.left), conv(x.right))
            raise ValueError(f"unsupported partition expr node {type(x).__name__}")

        return PartitionExpr(conv(e))

    def to_dict(self):
        return self.node

    @staticmethod
    def from_dict(d) -> "PartitionExpr":
        def fix(n):
            n = tuple(n) if isinstance(n, list) else n
            if n[0] in ("col", "lit"):
                return (n[0], n[1])
            return (n[0], fix(n[1]), fix(n[2]))
        return PartitionExpr(fix(d))

    def to_sql(self) -> str:
        def render(n):
            k = n[0]
            if k == "col":
                return n[1]
            if k == "lit":
                v = n[1]
                return f"'{v}'" if isinstance(v, str) else repr(v)
            sym = self._SQL[k]
            return f"{render(n[1])} {sym} {render(n[2])}"
        return render(self.node)

    # ------------------------------------------------------------- eval
    def columns(self) -> set:
        out = set()

        def walk(n):
            if n[0] == "col":
                out.add(n[1])
            elif n[0] != "lit":
                walk(n[1]); walk(n[2])
        walk(self.node)
        return out

    @staticmethod
    def _coerce(a, b):
        """SQL-ish comparison coercion: if one side is numeric and the
        other a numeric-looking string, compare numerically."""
        if isinstance(a, str) and isinstance(b, (int, float)):
            try:
                return float(a), float(b)
            except ValueError:
                return a, str(b)
        if isinstance(b, str) and isinstance(a, (int, float)):
            try:
                return float(a), float(b)
            except ValueError:
                return str(a), b
        return a, b

    def eval_row(self, values: dict) -> bool:
        def ev(n):
            k = n[0]
            if k == "and":
                return ev(n[1]) and ev(n[2])
            if k == "or":
                return ev(n[1]) or ev(n[2])
            lv = values.get(n[1][1]) if n[1][0] == "col" else n[1][1]
            rv = values.get(n[2][1]) if n[2][0] == "col" else n[2][1]
            if k == "eq":
                return lv == rv
            if k == "ne":
                return lv != rv
            if lv is None or rv is None:
                return False  # NULL fails range predicates (reference nulls_first ordering aside)
            lv, rv = self._coerce(lv, rv)
            return {"lt": lv < rv, "le": lv <= rv,
                    "gt": lv > rv, "ge": lv >= rv}[k]
        return bool(ev(self.node))

    def eval_vec(self, cols: dict[str, np.ndarray], n: int) -> np.ndarray:
        """Vectorized eval → bool[n] (reference split_record_batch)."""
        def ev(node):
            k = node[0]
            if k == "and":
                return ev(node[1]) & ev(node[2])
            if k == "or":
                return ev(node[1]) | ev(node[2])

            def side(s):
                if s[0] == "col":
                    c = cols.get(s[1])
                    if c is None:
                        return np.full(n, None, dtype=object)
                    return c
                return s[1]
            lv, rv = side(node[1]), side(node[2])
            if k == "eq":
                return np.asarray(lv == rv, dtype=bool) if not np.isscalar(lv) or not np.isscalar(rv) \
                    else np.full(n, lv == rv)
            if k == "ne":
                return np.asarray(lv != rv, dtype=bool)
            # range compares: mask out None, coerce strings vs numbers
            arr = lv if isinstance(lv, np.ndarray) else rv
            lit = rv if arr is lv else lv
            valid = np.array([v is not None for v in arr], dtype=bool)
            out = np.zeros(n, dtype=bool)
            if valid.any():
                av = arr[valid]
                if isinstance(lit, (int, float)):
                    try:
                        av = av.astype(np.float64)
                        litv = float(lit)
                    except (ValueError, TypeError):
                        av = av.astype(str)
                        litv = str(lit)
                else:
                    av = av.astype(str)
                    litv = str(lit)
                if arr is lv:
                    cm = {"lt": av < litv, "le": av <= litv,
                          "gt": av > litv, "ge": av >= litv}[k]
                else:  # literal on the left: lit OP col ≡ col invOP lit
                    cm = {"lt": av > litv, "le": av >= litv,
                          "gt": av < litv, "ge": av <= litv}[k]
                out[valid] = cm
            return out
        return ev(self.node)


class PartitionRule:
    """Base: maps series tag values → region index."""

    n_regions: int

    def region_of(self, values: dict) -> int:
        raise NotImplementedError

    def split(self, cols: dict[str, np.ndarray], n: int) -> np.ndarray:
        """Vectorized: region index per row (int32[n])."""
        raise NotImplementedError


class HashPartitionRule(PartitionRule):
    """Default rule when no PARTITION ON clause: hash(encoded pk) % n."""

    def __init__(self, n_regions: int, tag_names: list[str]):
        self.n_regions = n_regions
        self.tag_names = list(tag_names)

    def region_of(self, values: dict) -> int:
        from greptimedb_amd.engine import pk_codec
        pk = pk_codec.encode_pk(tuple(values.get(t) for t in self.tag_names))
        return tsid_hash(pk) % self.n_regions


class MultiDimPartitionRule(PartitionRule):
    """PARTITION ON COLUMNS (cols...) (expr0, expr1, ..., [default]).

    Row → first region whose expr matches; rows matching no expr go to the
    default region (an expr-less trailing slot, mirroring the reference's
    requirement that rules be exhaustive — we make them exhaustive by
    always keeping a catch-all region at the end)."""

    def __init__(self, columns: list[str], exprs: list[PartitionExpr | None]):
        self.columns = list(columns)
        self.exprs = list(exprs)
        if not self.exprs or self.exprs[-1] is not None:
            self.exprs.append(None)       # catch-all default region
        self.n_regions = len(self.exprs)

    def region_of(self, values: dict) -> int:
        for i, e in enumerate(self.exprs):
            if e is None or e.eval_row(values):
                return i
        return self.n_regions - 1  # unreachable (trailing None)

    def split(self, cols: dict[str, np.ndarray], n: int) -> np.ndarray:
        out = np.full(n, self.n_regions - 1, dtype=np.int32)
        unassigned = np.ones(n, dtype=bool)
        for i, e in enumerate(self.exprs):
            if e is None:
                continue
            m = unassigned & e.eval_vec(cols, n)
            out[m] = i
            unassigned &= ~m
        return out

    # ---------------------------------------------------------- persist
    def to_json(self) -> str:
        return json.dumps({
            "columns": self.columns,
            "exprs": [e.to_dict() if e is not None else None for e in self.exprs],
        })

    @staticmethod
    def from_json(s: str) -> "MultiDimPartitionRule":
        d = json.loads(s)
        exprs = [PartitionExpr.from_dict(e) if e is not None else None
                 for e in d["exprs"]]
        # from_json round-trips the stored trailing default — avoid doubling
        rule = MultiDimPartitionRule.__new__(MultiDimPartitionRule)
        rule.columns = list(d["columns"])
        rule.exprs = exprs
        if not rule.exprs or rule.exprs[-1] is not None:
            rule.exprs.append(None)
        rule.n_regions = len(rule.exprs)
        return rule

    def to_sql(self) -> str:
        parts = [e.to_sql() for e in self.exprs if e is not None]
        return (f"PARTITION ON COLUMNS ({', '.join(self.columns)}) "
                f"({', '.join(parts)})")


def rule_for_table(schema, n_regions: int) -> PartitionRule:
    """Resolve the table's partition rule: MultiDim when the schema options
    carry one, hash(pk) % n otherwise."""
    spec = schema.options.get("partition_rule")
    if spec:
        return MultiDimPartitionRule.from_json(spec)
    return HashPartitionRule(n_regions, [c.name for c in schema.tag_columns])
