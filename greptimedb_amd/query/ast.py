"""SQL AST node types (reference: src/sql statement types, trimmed to the
query surface implemented so far)."""

from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class Expr:
    pass


@dataclass
class Col(Expr):
    name: str


@dataclass
class Lit(Expr):
    value: object  # int | float | str | bool | None


@dataclass
class Interval(Expr):
    ms: int
    text: str


@dataclass
class Func(Expr):
    name: str
    args: list[Expr]
    distinct: bool = False


@dataclass
class Case(Expr):
    """CASE [operand] WHEN … THEN … [ELSE …] END (searched + simple forms)."""
    operand: Expr | None
    whens: list              # [(when_expr, result_expr)]
    default: Expr | None = None


@dataclass
class WindowFunc(Expr):
    """func(args) OVER (PARTITION BY … ORDER BY …) — default frame
    (RANGE UNBOUNDED PRECEDING..CURRENT ROW when ordered, whole partition
    otherwise; ref DataFusion window exprs)."""
    name: str
    args: list
    partition_by: list = field(default_factory=list)
    order_by: list = field(default_factory=list)   # (Expr, desc)


@dataclass
class ScalarSubquery(Expr):
    """(SELECT …) in an expression — uncorrelated; resolved to a literal
    (scalar) or value list (IN (...)) before planning the outer query."""
    select: object
    many: bool = False


@dataclass
class Exists(Expr):
    """EXISTS (SELECT …) — equality-correlated forms rewrite to IN;
    uncorrelated forms fold to a constant predicate at resolve time."""
    select: object
    negated: bool = False


@dataclass
class RangeAgg(Expr):
    """agg(col) RANGE '10s' [FILL x] — sliding window [t, t+range) per
    ALIGN step (ref: src/query/src/range_select/plan.rs:947 window math)."""
    func: "Func"
    range_ms: int
    fill: object = None        # None | "null" | "prev" | "linear" | float


@dataclass
class Star(Expr):
    pass


@dataclass
class CorrMap(Expr):
    """Decorrelated scalar subquery: per-outer-key scalar values.
    Evaluates to a per-row tensor via the outer tag column (LUT over series
    codes — the GPU-native join of `WHERE v > (SELECT agg .. WHERE inner.k
    = outer.k)`)."""
    map: dict          # key (str) -> float
    outer_col: str     # outer tag column supplying the key


@dataclass
class BinOp(Expr):
    op: str  # + - * / % = != < <= > >= and or like
    left: Expr
    right: Expr


@dataclass
class UnaryOp(Expr):
    op: str  # - not
    operand: Expr


@dataclass
class Cast(Expr):
    expr: Expr
    type: str  # lowered: bigint/int/double/float/string/timestamp/boolean


@dataclass
class SysVar(Expr):
    name: str  # @@name / @@session.name (lowered, @@ stripped)


@dataclass
class InList(Expr):
    expr: Expr
    items: list[Expr]
    negated: bool = False


@dataclass
class Between(Expr):
    expr: Expr
    low: Expr
    high: Expr
    negated: bool = False


@dataclass
class IsNull(Expr):
    expr: Expr
    negated: bool = False


@dataclass
class Join:
    table: str
    alias: str | None
    on: Expr                      # join condition (equality conjuncts)
    kind: str = "inner"           # inner | left


@dataclass
class Select:
    projections: list[tuple[Expr, str | None]]  # (expr, alias)
    table: str | None
    table_alias: str | None = None
    joins: list = field(default_factory=list)
    where: Expr | None = None
    group_by: list[Expr] = field(default_factory=list)
    having: Expr | None = None
    order_by: list[tuple[Expr, bool]] = field(default_factory=list)  # (expr, desc)
    limit: int | None = None
    offset: int | None = None
    # RANGE-query ALIGN clause (ref range_select): step, origin, by-cols, fill
    align_ms: int | None = None
    align_to: object = None            # None (epoch 0) | "now" | int ms | str
    align_by: list[str] | None = None
    align_fill: object = None
    distinct: bool = False             # SELECT DISTINCT
    ctes: list = field(default_factory=list)  # [(name, Select|SetOp)] WITH clause


@dataclass
class SetOp:
    """UNION [ALL] / EXCEPT / INTERSECT (reference setops cases)."""
    op: str                  # union | except | intersect
    all: bool
    left: object             # Select | SetOp
    right: object
    order_by: list = field(default_factory=list)
    limit: int | None = None
    ctes: list = field(default_factory=list)


@dataclass
class CreateView:
    name: str
    query_sql: str           # the view body, stored verbatim
    or_replace: bool = False
    if_not_exists: bool = False


@dataclass
class DropView:
    name: str
    if_exists: bool = False


@dataclass
class CreateTable:
    name: str
    columns: list[tuple[str, str, dict]]   # (name, type, {primary/time_index/null})
    primary_key: list[str]
    time_index: str | None
    if_not_exists: bool = False
    options: dict = field(default_factory=dict)
    partitions: int | None = None
    external: bool = False       # CREATE EXTERNAL TABLE (file engine)
    partition_on: tuple | None = None  # (columns, [Expr per region]) — PARTITION ON COLUMNS


@dataclass
class DropTable:
    name: str
    if_exists: bool = False


@dataclass
class TruncateTable:
    name: str


@dataclass
class DeclareCursor:
    name: str
    select: object


@dataclass
class FetchCursor:
    name: str
    count: int


@dataclass
class CloseCursor:
    name: str


@dataclass
class ShowTableStatus:
    like: str | None = None


@dataclass
class ValuesTable:
    rows: list
    columns: list | None = None


@dataclass
class ShowTables:
    like: str | None = None
    full: bool = False


@dataclass
class ShowDatabases:
    like: str | None = None


@dataclass
class ShowCreateView:
    name: str


@dataclass
class ShowCreateFlow:
    name: str


@dataclass
class ShowIndex:
    name: str


@dataclass
class ShowCreateTable:
    name: str


@dataclass
class DescribeTable:
    name: str


@dataclass
class InsertValues:
    table: str
    columns: list[str]
    rows: list[list[object]]
    select: object = None   # INSERT INTO t [cols] SELECT ... (rows empty)


@dataclass
class Delete:
    table: str
    where: object | None


@dataclass
class AlterTable:
    table: str
    action: str        # add_column | set_options | unset_options | rename
    column: tuple | None = None  # (name, type, opts)
    options: dict | None = None  # for set/unset/rename


@dataclass
class Copy:
    table: str
    path: str
    direction: str  # "to" | "from"
    options: dict = field(default_factory=dict)


@dataclass
class CreateFlow:
    name: str
    sink: str
    query_sql: str
    if_not_exists: bool = False
    expire_after_s: int | None = None   # EXPIRE AFTER '1h' — dirty-window TTL


@dataclass
class DropFlow:
    name: str


@dataclass
class ShowFlows:
    pass


@dataclass
class Admin:
    """ADMIN func(args) — flush_table / compact_table / flush_region ...
    (reference: src/common/function admin functions)."""
    func: str
    args: list


@dataclass
class Explain:
    analyze: bool
    stmt: object


@dataclass
class Tql:
    """TQL EVAL (start, end, step) promql_expr  (reference: src/sql TQL)."""
    start: float
    end: float
    step: float
    query: str


@dataclass
class Use:
    schema: str


@dataclass
class SetVar:
    name: str
    value: object


@dataclass
class CreateDatabase:
    name: str
    if_not_exists: bool = False


@dataclass
class DropDatabase:
    name: str
    if_exists: bool = False


@dataclass
class ShowVariables:
    like: str | None = None


@dataclass
class Kill:
    pid: int
