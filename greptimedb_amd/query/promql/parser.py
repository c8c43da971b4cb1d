"""PromQL parser (hand-written; reference uses the promql-parser crate).

Covers the practical query surface: selectors with matchers and [range]
/ offset / @ (timestamp or start()/end()), subqueries [range:res],
functions, aggregations with by/without, arithmetic/comparison binary
operators with precedence, unary minus, parentheses.
"""

from __future__ import annotations

import re

from greptimedb_amd.query.promql import ast
from greptimedb_amd.utils.errors import InvalidSyntax

_TOK = re.compile(r"""
    \s+
  | (?P<dur>\d+(?:\.\d+)?(?:ms|s|m|h|d|w|y)(?:\d+(?:\.\d+)?(?:ms|s|m|h|d|w|y))*)
  | (?P<num>0x[0-9a-fA-F]+|\d+\.\d*(?:[eE][+-]?\d+)?|\.\d+(?:[eE][+-]?\d+)?|\d+(?:[eE][+-]?\d+)?|[Ii]nf|NaN)
  | (?P<str>"(?:\\.|[^"\\])*"|'(?:\\.|[^'\\])*')
  | (?P<id>[a-zA-Z_:][a-zA-Z0-9_:]*)
  | (?P<op>=~|!~|!=|==|<=|>=|<|>|\+|-|\*|/|%|\^|\(|\)|\{|\}|\[|\]|,|=|@)
""", re.VERBOSE)

_UNIT_S = {"ms": 0.001, "s": 1, "m": 60, "h": 3600, "d": 86400, "w": 604800, "y": 31536000}

_AGG_OPS = {"sum", "avg", "min", "max", "count", "group", "stddev", "stdvar",
            "topk", "bottomk", "quantile", "count_values"}
_PARAM_AGGS = {"topk", "bottomk", "quantile", "count_values"}

_CMP = {"==", "!=", "<", "<=", ">", ">="}
_PREC = {"or": 1, "unless": 2, "and": 2,
         "==": 3, "!=": 3, "<": 3, "<=": 3, ">": 3, ">=": 3,
         "+": 4, "-": 4, "*": 5, "/": 5, "%": 5, "^": 6}


def parse_duration_s(text: str) -> float:
    total = 0.0
    for num, unit in re.findall(r"(\d+(?:\.\d+)?)(ms|s|m|h|d|w|y)", text):
        total += float(num) * _UNIT_S[unit]
    return total


def tokenize(q: str):
    out = []
    pos = 0
    while pos < len(q):
        m = _TOK.match(q, pos)
        if not m:
            raise InvalidSyntax(f"promql: bad token at {q[pos:pos+15]!r}")
        pos = m.end()
        kind = m.lastgroup
        if kind is None:
            continue
        v = m.group()
        if kind == "dur":
            out.append(("dur", parse_duration_s(v)))
        elif kind == "num":
            out.append(("num", float(int(v, 16)) if v.startswith("0x") else
                        float("inf") if v.lower() == "inf" else float(v)))
        elif kind == "str":
            out.append(("str", v[1:-1].encode().decode("unicode_escape")))
        elif kind == "id":
            out.append(("id", v))
        else:
            out.append(("op", v))
    return out


class PromParser:
    def __init__(self, q: str):
        self.toks = tokenize(q)
        self.i = 0

    def peek(self):
        return self.toks[self.i] if self.i < len(self.toks) else (None, None)

    def next(self):
        t = self.peek()
        if t[0] is None:
            raise InvalidSyntax("promql: unexpected end")
        self.i += 1
        return t

    def eat_op(self, op):
        if self.peek() == ("op", op):
            self.i += 1
            return True
        return False

    def expect_op(self, op):
        if not self.eat_op(op):
            raise InvalidSyntax(f"promql: expected {op!r} near {self.peek()}")

    def parse(self):
        e = self.parse_expr(0)
        if self.peek()[0] is not None:
            raise InvalidSyntax(f"promql: trailing {self.peek()}")
        return e

    def parse_expr(self, min_prec):
        left = self.parse_atom()
        while True:
            k, v = self.peek()
            opname = None
            if k == "op" and v in _PREC:
                opname = v
            elif k == "id" and v in ("and", "or", "unless"):
                opname = v
            if opname is None or _PREC[opname] <= min_prec:
                break
            self.i += 1
            bool_mod = False
            on = ignoring = None
            gl = gr = False
            if self.peek() == ("id", "bool"):
                self.i += 1
                bool_mod = True
            if self.peek()[0] == "id" and self.peek()[1] in ("on", "ignoring"):
                which = self.next()[1]
                labels = self._label_list()
                if which == "on":
                    on = labels
                else:
                    ignoring = labels
                if self.peek()[0] == "id" and self.peek()[1] in ("group_left", "group_right"):
                    w2 = self.next()[1]
                    gl = w2 == "group_left"
                    gr = not gl
                    if self.peek() == ("op", "("):
                        self._label_list()
            right = self.parse_expr(_PREC[opname])
            left = ast.BinOp(opname, left, right, bool_mod, on, ignoring, gl, gr)
        return left

    def _label_list(self):
        self.expect_op("(")
        out = []
        while not self.eat_op(")"):
            k, v = self.next()
            if k == "id" or k == "str":
                out.append(v)
            self.eat_op(",")
        return out

    def parse_atom(self):
        k, v = self.peek()
        if k == "num":
            self.i += 1
            return ast.NumberLit(v)
        if k == "str":
            self.i += 1
            return ast.StringLit(v)
        if k == "op" and v == "-":
            self.i += 1
            return ast.Unary("-", self.parse_atom())
        if k == "op" and v == "+":
            self.i += 1
            return self.parse_atom()
        if k == "op" and v == "(":
            self.i += 1
            e = self.parse_expr(0)
            self.expect_op(")")
            return self._postfix(e)
        if k == "op" and v == "{":
            return self._postfix(self._selector(None))
        if k == "id":
            self.i += 1
            # aggregation?
            if v in _AGG_OPS and self.peek()[0] == "op" and self.peek()[1] == "(" or \
               v in _AGG_OPS and self.peek() == ("id", "by") or \
               v in _AGG_OPS and self.peek() == ("id", "without"):
                return self._postfix(self._aggregate(v))
            if self.peek() == ("op", "("):
                # function call
                self.i += 1
                args = []
                while not self.eat_op(")"):
                    args.append(self.parse_expr(0))
                    self.eat_op(",")
                return self._postfix(ast.Call(v, args))
            return self._postfix(self._selector(v))
        raise InvalidSyntax(f"promql: unexpected {k}:{v}")

    def _aggregate(self, op):
        by = without = None
        if self.peek() == ("id", "by"):
            self.i += 1
            by = self._label_list()
        elif self.peek() == ("id", "without"):
            self.i += 1
            without = self._label_list()
        self.expect_op("(")
        first = self.parse_expr(0)
        param = None
        if self.eat_op(","):
            param, first = first, self.parse_expr(0)
        self.expect_op(")")
        if self.peek() == ("id", "by"):
            self.i += 1
            by = self._label_list()
        elif self.peek() == ("id", "without"):
            self.i += 1
            without = self._label_list()
        return ast.Aggregate(op, first, by, without, param)

    def _selector(self, metric):
        matchers = []
        if self.eat_op("{"):
            while not self.eat_op("}"):
                name = self.next()[1]
                k, op = self.next()
                if op not in ("=", "!=", "=~", "!~"):
                    raise InvalidSyntax(f"promql: bad matcher op {op}")
                val = self.next()
                if val[0] != "str":
                    raise InvalidSyntax("promql: matcher value must be string")
                matchers.append(ast.Matcher(name, op, val[1]))
                self.eat_op(",")
        sel = ast.Selector(metric, matchers)
        return self._postfix(sel)

    def _postfix(self, e):
        # [range], [range:resolution] (subquery) and offset
        while True:
            if self.eat_op("["):
                k, v = self.next()
                if k != "dur":
                    raise InvalidSyntax("promql: expected duration in [...]")
                # subquery: the ':' lexes as an id token (metric names may
                # contain colons), possibly fused with the resolution
                nk, nv = self.peek()
                if nk == "id" and nv.startswith(":"):
                    self.i += 1
                    res = 0.0
                    rest = nv[1:]
                    if rest:
                        if not re.fullmatch(
                                r"(?:\d+(?:\.\d+)?(?:ms|s|m|h|d|w|y))+", rest):
                            raise InvalidSyntax(
                                f"promql: bad subquery resolution {rest!r}")
                        res = parse_duration_s(rest)
                    elif self.peek()[0] == "dur":
                        res = self.next()[1]
                    self.expect_op("]")
                    e = ast.Subquery(e, v, res)
                    continue
                if not isinstance(e, ast.Selector):
                    raise InvalidSyntax("promql: range on non-selector "
                                        "(use [range:step] for a subquery)")
                e.range_s = v
                self.expect_op("]")
            elif self.peek() == ("id", "offset"):
                self.i += 1
                k, v = self.next()
                if k != "dur":
                    raise InvalidSyntax("promql: expected duration after offset")
                if isinstance(e, (ast.Selector, ast.Subquery)):
                    e.offset_s = v
                else:
                    raise InvalidSyntax("promql: offset on non-selector")
            elif self.peek() == ("op", "@"):
                self.i += 1
                k, v = self.next()
                if k == "num":
                    at = float(v)
                elif k == "id" and v in ("start", "end"):
                    self.expect_op("(")
                    self.expect_op(")")
                    at = v                     # resolved at eval time
                else:
                    raise InvalidSyntax("promql: @ needs a timestamp or start()/end()")
                if isinstance(e, (ast.Selector, ast.Subquery)):
                    e.at_s = at
                else:
                    raise InvalidSyntax("promql: @ on non-selector")
            else:
                return e


def parse_promql(q: str):
    return PromParser(q).parse()
