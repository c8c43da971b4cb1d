"""PromQL AST (reference: promql-parser crate + src/query/src/promql/planner.rs)."""

from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class Matcher:
    name: str
    op: str        # = != =~ !~
    value: str


@dataclass
class Selector:
    metric: str | None
    matchers: list[Matcher] = field(default_factory=list)
    range_s: float | None = None     # [5m] window (seconds); None = instant vector
    offset_s: float = 0.0
    at_s: object = None              # @ modifier: epoch s | "start" | "end"


@dataclass
class Subquery:
    """expr[range:resolution] — inner expr evaluated on its own grid, the
    results treated as samples for the enclosing range function
    (Prometheus SubqueryExpr; step_s 0 = default resolution)."""
    expr: object
    range_s: float
    step_s: float = 0.0
    offset_s: float = 0.0
    at_s: object = None              # @ modifier


@dataclass
class NumberLit:
    value: float


@dataclass
class StringLit:
    value: str


@dataclass
class Call:
    func: str
    args: list


@dataclass
class Aggregate:
    op: str                     # sum avg min max count topk bottomk quantile ...
    expr: object
    by: list[str] | None = None       # by(...) labels
    without: list[str] | None = None
    param: object | None = None       # topk(k, ...) / quantile(q, ...)


@dataclass
class BinOp:
    op: str
    left: object
    right: object
    bool_modifier: bool = False
    on: list[str] | None = None
    ignoring: list[str] | None = None
    group_left: bool = False
    group_right: bool = False


@dataclass
class Unary:
    op: str
    expr: object
