"""Minimal VRL (Vector Remap Language) interpreter for the `vrl` processor.

Reference parity: src/pipeline/src/etl/processor/vrl.rs runs a VRL program
against each pipeline row. This implements the remap subset that covers
the documented pipeline use cases:

  statements:  `.path = expr` assignments, `del(.path)`, `if cond { ... }
               [else { ... }]`
  exprs:       `.path` field refs, string/number/bool/null literals,
               `+ - * /` arithmetic, string `+` concat, comparisons,
               `&&`/`||`/`!`, and functions: upcase, downcase, to_string,
               to_int, to_float, length, contains, starts_with, ends_with,
               replace, split, join, trim, sha256, now, exists

The program is parsed once per pipeline (AST cached) and evaluated per
row against a dict. Unknown syntax raises at pipeline-creation time, like
the reference compiling the VRL program up front.
"""

from __future__ import annotations

import hashlib
import re
import time

_TOKEN = re.compile(r"""
    \s+
  | (?P<comment>\#[^\n]*)
  | (?P<num>\d+\.\d+|\d+)
  | (?P<str>"(?:[^"\\]|\\.)*")
  | (?P<path>\.[A-Za-z_][A-Za-z0-9_.]*)
  | (?P<id>[A-Za-z_][A-Za-z0-9_]*)
  | (?P<op>==|!=|<=|>=|&&|\|\||[=+\-*/<>!(){},;])
""", re.VERBOSE)


def _tokenize(src: str):
    out, pos = [], 0
    while pos < len(src):
        m = _TOKEN.match(src, pos)
        if not m:
            raise ValueError(f"vrl: bad token at {src[pos:pos+20]!r}")
        pos = m.end()
        kind = m.lastgroup
        if kind is None or kind == "comment":
            continue
        v = m.group()
        if kind == "num":
            out.append(("num", float(v) if "." in v else int(v)))
        elif kind == "str":
            out.append(("str", v[1:-1].replace('\\"', '"').replace("\\n", "\n")))
        else:
            out.append((kind, v))
    return out


class _Parser:
    def __init__(self, toks):
        self.toks = toks
        self.i = 0

    def peek(self):
        return self.toks[self.i] if self.i < len(self.toks) else (None, None)

    def next(self):
        t = self.peek()
        self.i += 1
        return t

    def eat(self, kind, val=None):
        k, v = self.peek()
        if k == kind and (val is None or v == val):
            self.i += 1
            return True
        return False

    def expect(self, kind, val=None):
        if not self.eat(kind, val):
            raise ValueError(f"vrl: expected {val or kind} near {self.peek()}")

    def program(self):
        stmts = []
        while self.peek()[0] is not None and self.peek() != ("op", "}"):
            stmts.append(self.statement())
            while self.eat("op", ";"):
                pass
        return stmts

    def statement(self):
        k, v = self.peek()
        if k == "id" and v == "del":
            self.next()
            self.expect("op", "(")
            _k, path = self.next()
            self.expect("op", ")")
            return ("del", path[1:])
        if k == "id" and v == "if":
            self.next()
            cond = self.expr()
            self.expect("op", "{")
            then = self.program()
            self.expect("op", "}")
            els = []
            if self.eat("id", "else"):
                self.expect("op", "{")
                els = self.program()
                self.expect("op", "}")
            return ("if", cond, then, els)
        if k == "path":
            self.next()
            self.expect("op", "=")
            return ("set", v[1:], self.expr())
        raise ValueError(f"vrl: unexpected {k} {v!r}")

    def expr(self, prec=0):
        left = self.unary()
        OPS = {"||": 1, "&&": 2, "==": 3, "!=": 3, "<": 3, "<=": 3,
               ">": 3, ">=": 3, "+": 4, "-": 4, "*": 5, "/": 5}
        while True:
            k, v = self.peek()
            if k != "op" or v not in OPS or OPS[v] <= prec:
                return left
            self.next()
            left = ("bin", v, left, self.expr(OPS[v]))

    def unary(self):
        k, v = self.peek()
        if k == "op" and v == "!":
            self.next()
            return ("not", self.unary())
        if k == "op" and v == "(":
            self.next()
            e = self.expr()
            self.expect("op", ")")
            return e
        if k == "num" or k == "str":
            self.next()
            return ("lit", v)
        if k == "path":
            self.next()
            return ("ref", v[1:])
        if k == "id":
            self.next()
            if v in ("true", "false"):
                return ("lit", v == "true")
            if v == "null":
                return ("lit", None)
            self.expect("op", "(")
            args = []
            if not self.eat("op", ")"):
                while True:
                    args.append(self.expr())
                    if not self.eat("op", ","):
                        break
                self.expect("op", ")")
            return ("call", v, args)
        raise ValueError(f"vrl: unexpected {k} {v!r}")


_FUNCS = {
    "upcase": lambda a: None if a[0] is None else str(a[0]).upper(),
    "downcase": lambda a: None if a[0] is None else str(a[0]).lower(),
    "to_string": lambda a: None if a[0] is None else str(a[0]),
    "to_int": lambda a: None if a[0] is None else int(float(a[0])),
    "to_float": lambda a: None if a[0] is None else float(a[0]),
    "length": lambda a: 0 if a[0] is None else len(a[0]),
    "contains": lambda a: a[0] is not None and str(a[1]) in str(a[0]),
    "starts_with": lambda a: a[0] is not None and str(a[0]).startswith(str(a[1])),
    "ends_with": lambda a: a[0] is not None and str(a[0]).endswith(str(a[1])),
    "replace": lambda a: None if a[0] is None else
        str(a[0]).replace(str(a[1]), str(a[2])),
    "split": lambda a: [] if a[0] is None else str(a[0]).split(str(a[1])),
    "join": lambda a: str(a[1]).join(str(x) for x in (a[0] or [])),
    "trim": lambda a: None if a[0] is None else str(a[0]).strip(),
    "sha256": lambda a: hashlib.sha256(str(a[0]).encode()).hexdigest(),
    "now": lambda a: int(time.time() * 1000),
}


class VrlProgram:
    def __init__(self, source: str):
        self.stmts = _Parser(_tokenize(source)).program()

    # --------------------------------------------------------------- eval
    def run(self, row: dict) -> dict:
        self._exec(self.stmts, row)
        return row

    def _exec(self, stmts, row):
        for st in stmts:
            if st[0] == "set":
                row[st[1]] = self._eval(st[2], row)
            elif st[0] == "del":
                row.pop(st[1], None)
            elif st[0] == "if":
                if self._truthy(self._eval(st[1], row)):
                    self._exec(st[2], row)
                else:
                    self._exec(st[3], row)

    @staticmethod
    def _truthy(v):
        return bool(v)

    def _eval(self, e, row):
        kind = e[0]
        if kind == "lit":
            return e[1]
        if kind == "ref":
            return row.get(e[1])
        if kind == "not":
            return not self._truthy(self._eval(e[1], row))
        if kind == "call":
            fn = _FUNCS.get(e[1])
            if fn is None:
                if e[1] == "exists":
                    return e[2][0][1] in row if e[2][0][0] == "ref" else False
                raise ValueError(f"vrl: unknown function {e[1]}")
            return fn([self._eval(a, row) for a in e[2]])
        if kind == "bin":
            op = e[1]
            l = self._eval(e[2], row)
            if op == "&&":
                return self._truthy(l) and self._truthy(self._eval(e[3], row))
            if op == "||":
                return self._truthy(l) or self._truthy(self._eval(e[3], row))
            r = self._eval(e[3], row)
            if op == "+":
                if isinstance(l, str) or isinstance(r, str):
                    return ("" if l is None else str(l)) + \
                        ("" if r is None else str(r))
                return (l or 0) + (r or 0)
            if l is None or r is None:
                return {"==": l == r, "!=": l != r}.get(op, False)
            return {"-": lambda: l - r, "*": lambda: l * r,
                    "/": lambda: l / r if r else None,
                    "==": lambda: l == r, "!=": lambda: l != r,
                    "<": lambda: l < r, "<=": lambda: l <= r,
                    ">": lambda: l > r, ">=": lambda: l >= r}[op]()
        raise ValueError(f"vrl: bad node {e}")
