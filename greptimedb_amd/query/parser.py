"""Recursive-descent SQL parser for the supported dialect.

Reference parity: src/sql (sqlparser-rs based). We implement a hand-written
tokenizer + Pratt expression parser covering the query/DDL/DML surface the
engine executes: SELECT (WHERE/GROUP BY/HAVING/ORDER BY/LIMIT), CREATE
TABLE (TIME INDEX, PRIMARY KEY, WITH options, PARTITION), DROP, SHOW,
DESCRIBE, INSERT VALUES, TQL EVAL.
"""

from __future__ import annotations

import re

from greptimedb_amd.query import ast
from greptimedb_amd.utils.errors import InvalidSyntax


def _duration_ms(text: str) -> int:
    """'10s' / '5m' / '1h30m' → milliseconds (RANGE/ALIGN durations)."""
    from greptimedb_amd.query.promql.parser import parse_duration_s
    ms = int(parse_duration_s(text) * 1000)
    if ms <= 0:
        raise InvalidSyntax(f"bad duration {text!r}")
    return ms

_TOKEN_RE = re.compile(r"""
    \s+
  | (?P<comment>--[^\n]*)
  | (?P<num>\d+\.\d*(?:[eE][+-]?\d+)?|\.\d+(?:[eE][+-]?\d+)?|\d+(?:[eE][+-]?\d+)?)
  | (?P<str>'(?:[^']|'')*')
  | (?P<qid>"(?:[^"]|"")*")
  | (?P<id>[A-Za-z_][A-Za-z0-9_.]*)
  | (?P<op>=~|!~|<>|!=|<=|>=|=|<|>|\(|\)|\[|\]|\{|\}|,|\*|\+|-|/|%|;|::|:|@)
""", re.VERBOSE)

_UNITS_MS = {
    "millisecond": 1, "milliseconds": 1, "ms": 1,
    "second": 1000, "seconds": 1000, "s": 1000,
    "minute": 60_000, "minutes": 60_000, "m": 60_000,
    "hour": 3_600_000, "hours": 3_600_000, "h": 3_600_000,
    "day": 86_400_000, "days": 86_400_000, "d": 86_400_000,
    "week": 604_800_000, "weeks": 604_800_000,
}


def parse_interval_text(text: str) -> int:
    """'1 minute' / '5m' / '1 hour 30 minutes' → ms."""
    total = 0
    for num, unit in re.findall(r"([\d.]+)\s*([A-Za-z]+)", text):
        u = unit.lower()
        if u not in _UNITS_MS:
            raise InvalidSyntax(f"unknown interval unit {unit!r}")
        total += float(num) * _UNITS_MS[u]
    if total == 0 and text.strip():
        try:
            total = float(text) * 1000  # bare seconds
        except ValueError:
            raise InvalidSyntax(f"bad interval {text!r}") from None
    return int(total)


class Token:
    __slots__ = ("kind", "value", "pos")

    def __init__(self, kind, value, pos=0):
        self.kind = kind
        self.value = value
        self.pos = pos

    def __repr__(self):
        return f"{self.kind}:{self.value}"


def tokenize(sql: str) -> list[Token]:
    out = []
    pos = 0
    while pos < len(sql):
        m = _TOKEN_RE.match(sql, pos)
        if not m:
            raise InvalidSyntax(f"bad token at {sql[pos:pos+20]!r}")
        start = m.start()
        pos = m.end()
        if m.lastgroup is None or m.lastgroup == "comment":
            continue
        kind = m.lastgroup
        v = m.group()
        if kind == "str":
            out.append(Token("str", v[1:-1].replace("''", "'"), start))
        elif kind == "qid":
            out.append(Token("id", v[1:-1].replace('""', '"'), start))
        elif kind == "num":
            out.append(Token("num", float(v) if ("." in v or "e" in v or "E" in v) else int(v), start))
        elif kind == "id":
            out.append(Token("id", v, start))
        else:
            out.append(Token("op", v, start))
    return out


_PRECEDENCE = {
    "or": 1, "and": 2,
    "=": 4, "!=": 4, "<>": 4, "<": 4, "<=": 4, ">": 4, ">=": 4,
    "like": 4, "in": 4, "between": 4, "is": 4,
    "+": 5, "-": 5, "*": 6, "/": 6, "%": 6,
}


class Parser:
    def __init__(self, sql: str):
        self.sql = sql
        self.toks = tokenize(sql)
        self.i = 0

    # ---------------- token helpers ----------------
    def peek(self) -> Token | None:
        return self.toks[self.i] if self.i < len(self.toks) else None

    def next(self) -> Token:
        t = self.peek()
        if t is None:
            raise InvalidSyntax("unexpected end of query")
        self.i += 1
        return t

    def at_kw(self, *kws) -> bool:
        t = self.peek()
        return t is not None and t.kind == "id" and t.value.lower() in kws

    def eat_kw(self, *kws) -> bool:
        if self.at_kw(*kws):
            self.i += 1
            return True
        return False

    def expect_kw(self, kw):
        if not self.eat_kw(kw):
            raise InvalidSyntax(f"expected {kw.upper()} near {self.peek()}")

    def at_op(self, op) -> bool:
        t = self.peek()
        return t is not None and t.kind == "op" and t.value == op

    def eat_op(self, op) -> bool:
        if self.at_op(op):
            self.i += 1
            return True
        return False

    def expect_op(self, op):
        if not self.eat_op(op):
            raise InvalidSyntax(f"expected {op!r} near {self.peek()}")

    # ---------------- statements ----------------
    def parse_statement(self):
        stmt = self.parse_statement_inner()
        self.eat_op(";")
        if self.peek() is not None:
            raise InvalidSyntax(f"trailing tokens: {self.peek()}")
        return stmt

    def parse_statement_inner(self):
        if self.at_kw("select", "with"):
            stmt = self.parse_query()
        elif self.at_kw("use"):
            self.next()
            stmt = ast.Use(self.next().value)
        elif self.at_kw("kill"):
            self.next()
            self.eat_kw("query")
            stmt = ast.Kill(int(self.next().value))
        elif self.at_kw("set"):
            self.next()
            self.eat_kw("session") or self.eat_kw("global")
            name = str(self.next().value)
            self.expect_op("=")
            t = self.next()
            stmt = ast.SetVar(name, t.value)
        elif self.at_kw("create"):
            stmt = self.parse_create()
        elif self.at_kw("drop"):
            stmt = self.parse_drop()
        elif self.at_kw("truncate"):
            self.next()
            self.eat_kw("table")
            stmt = ast.TruncateTable(str(self.next().value))
        elif self.at_kw("declare"):
            # DECLARE c CURSOR FOR SELECT … (ref sql statements/cursor.rs)
            self.next()
            name = str(self.next().value)
            self.expect_kw("cursor")
            self.expect_kw("for")
            stmt = ast.DeclareCursor(name, self.parse_query())
        elif self.at_kw("fetch"):
            self.next()
            n = 1
            t = self.peek()
            if t is not None and t.kind == "num":
                n = int(self.next().value)
            self.eat_kw("from")
            stmt = ast.FetchCursor(str(self.next().value), n)
        elif self.at_kw("close"):
            self.next()
            stmt = ast.CloseCursor(str(self.next().value))
        elif self.at_kw("show"):
            self.next()
            if self.eat_kw("flows"):
                stmt = ast.ShowFlows()
            elif self.eat_kw("databases") or self.eat_kw("schemas"):
                like = None
                if self.eat_kw("like"):
                    like = str(self.next().value)
                stmt = ast.ShowDatabases(like)
            elif self.eat_kw("create"):
                if self.eat_kw("view"):
                    stmt = ast.ShowCreateView(self.next().value)
                elif self.eat_kw("flow"):
                    stmt = ast.ShowCreateFlow(self.next().value)
                else:
                    self.expect_kw("table")
                    stmt = ast.ShowCreateTable(self.next().value)
            elif self.eat_kw("columns") or self.eat_kw("fields"):
                self.expect_kw("from")
                stmt = ast.DescribeTable(self.next().value)
            elif self.eat_kw("index", "indexes"):
                self.expect_kw("from")
                stmt = ast.ShowIndex(self.next().value)
            elif self.eat_kw("variables"):
                like = None
                if self.eat_kw("like"):
                    like = str(self.next().value)
                stmt = ast.ShowVariables(like)
            elif self.eat_kw("table"):
                self.expect_kw("status")
                like = None
                if self.eat_kw("like"):
                    like = str(self.next().value)
                stmt = ast.ShowTableStatus(like)
            else:
                full = self.eat_kw("full")
                self.expect_kw("tables")
                like = None
                if self.eat_kw("like"):
                    like = str(self.next().value)
                stmt = ast.ShowTables(like, full)
        elif self.at_kw("describe", "desc"):
            self.next()
            self.eat_kw("table")
            stmt = ast.DescribeTable(self.next().value)
        elif self.at_kw("insert"):
            stmt = self.parse_insert()
        elif self.at_kw("tql"):
            stmt = self.parse_tql()
        elif self.at_kw("admin"):
            self.next()
            e = self.parse_expr()
            if not isinstance(e, ast.Func):
                raise InvalidSyntax("ADMIN expects a function call")
            args = [a.value if isinstance(a, ast.Lit) else a for a in e.args]
            stmt = ast.Admin(e.name, args)
        elif self.at_kw("explain"):
            self.next()
            analyze = self.eat_kw("analyze")
            stmt = ast.Explain(analyze, self.parse_statement_inner())
        elif self.at_kw("delete"):
            self.next()
            self.expect_kw("from")
            table = self.next().value
            where = self.parse_expr() if self.eat_kw("where") else None
            stmt = ast.Delete(table, where)
        elif self.at_kw("alter"):
            self.next()
            self.expect_kw("table")
            table = self.next().value
            if self.eat_kw("set"):
                opts = {}
                while True:
                    k = str(self.next().value)
                    self.expect_op("=")
                    opts[k.lower()] = str(self.next().value)
                    if not self.eat_op(","):
                        break
                return ast.AlterTable(table, "set_options", options=opts)
            if self.eat_kw("unset"):
                keys = []
                while True:
                    keys.append(str(self.next().value).lower())
                    if not self.eat_op(","):
                        break
                return ast.AlterTable(table, "unset_options",
                                      options={k: None for k in keys})
            if self.eat_kw("rename"):
                self.eat_kw("to")
                return ast.AlterTable(table, "rename",
                                      options={"to": str(self.next().value)})
            self.expect_kw("add")
            self.eat_kw("column")
            cname = self.next().value
            ctype = self.next().value
            if self.eat_op("("):
                ctype += f"({self.next().value})"
                self.expect_op(")")
            opts = {}
            if self.eat_kw("fulltext"):
                self.eat_kw("index")
                opts["fulltext"] = True
            stmt = ast.AlterTable(table, "add_column", (cname, ctype, opts))
        elif self.at_kw("copy"):
            self.next()
            table = self.next().value
            if self.eat_kw("to"):
                direction = "to"
            elif self.eat_kw("from"):
                direction = "from"
            else:
                raise InvalidSyntax("COPY <table> TO|FROM '<path>'")
            path = self.next().value
            options = {}
            # WITH (format=...) and CONNECTION (endpoint=..., ...) — the
            # reference's COPY grammar (sql/src/parsers/copy_parser.rs);
            # connection keys merge into one option map here
            while self.at_kw("with") or self.at_kw("connection"):
                self.next()
                self.expect_op("(")
                while not self.eat_op(")"):
                    k = self.next().value
                    self.expect_op("=")
                    options[str(k).lower()] = str(self.next().value)
                    self.eat_op(",")
            stmt = ast.Copy(table, str(path), direction, options)
        else:
            raise InvalidSyntax(f"unsupported statement start: {self.peek()}")
        return stmt

    def parse_query(self):
        """[WITH ctes] select [UNION/EXCEPT/INTERSECT select ...] — a
        trailing ORDER BY/LIMIT binds to the whole set operation."""
        ctes = []
        if self.eat_kw("with"):
            while True:
                name = self.next().value
                self.expect_kw("as")
                self.expect_op("(")
                sub = self.parse_query()
                self.expect_op(")")
                ctes.append((name, sub))
                if not self.eat_op(","):
                    break
        node = self.parse_select()
        while self.at_kw("union", "except", "intersect"):
            op = self.next().value.lower()
            all_rows = self.eat_kw("all")
            self.eat_kw("distinct")
            right = self.parse_select()
            node = ast.SetOp(op, all_rows, node, right)
        if isinstance(node, ast.SetOp):
            rs = node.right
            if isinstance(rs, ast.Select) and (rs.order_by or
                                               rs.limit is not None):
                node.order_by, node.limit = rs.order_by, rs.limit
                rs.order_by, rs.limit = [], None
        if ctes:
            node.ctes = ctes
        return node

    def parse_select(self) -> ast.Select:
        self.expect_kw("select")
        distinct = self.eat_kw("distinct")
        projections = []
        while True:
            if self.eat_op("*"):
                projections.append((ast.Star(), None))
            else:
                e = self.parse_expr()
                alias = None
                if self.eat_kw("as"):
                    alias = self.next().value
                elif self.peek() is not None and self.peek().kind == "id" and \
                        self.peek().value.lower() not in (
                            "from", "where", "group", "order", "limit", "having",
                            "offset", "align", "fill", "range",
                            "union", "except", "intersect"):
                    alias = self.next().value
                projections.append((e, alias))
            if not self.eat_op(","):
                break
        table = None
        table_alias = None
        joins = []
        if self.eat_kw("from"):
            if self.at_op("("):
                self.next()
                if self.at_kw("values"):
                    table = self._parse_values_tail()
                else:
                    table = self.parse_query()   # derived table
                self.expect_op(")")
            else:
                table = self.next().value
            if self.eat_kw("as"):
                table_alias = self.next().value
                if isinstance(table, ast.ValuesTable) and self.eat_op("("):
                    cols = []
                    while not self.eat_op(")"):
                        cols.append(str(self.next().value))
                        self.eat_op(",")
                    table.columns = cols
            elif self.peek() is not None and self.peek().kind == "id" and \
                    self.peek().value.lower() not in (
                        "where", "group", "order", "limit", "having", "offset",
                        "join", "inner", "left", "on", "align",
                        "union", "except", "intersect"):
                table_alias = self.next().value
            while self.at_kw("join", "inner", "left"):
                kind = "inner"
                if self.eat_kw("left"):
                    kind = "left"
                    self.eat_kw("outer")
                else:
                    self.eat_kw("inner")
                self.expect_kw("join")
                jt = self.next().value
                jalias = None
                if self.eat_kw("as"):
                    jalias = self.next().value
                elif self.peek() is not None and self.peek().kind == "id" and \
                        self.peek().value.lower() != "on":
                    jalias = self.next().value
                self.expect_kw("on")
                on = self.parse_expr()
                joins.append(ast.Join(jt, jalias, on, kind))
        where = None
        if self.eat_kw("where"):
            where = self.parse_expr()
        group_by = []
        if self.eat_kw("group"):
            self.expect_kw("by")
            while True:
                group_by.append(self.parse_expr())
                if not self.eat_op(","):
                    break
        having = None
        if self.eat_kw("having"):
            having = self.parse_expr()
        # ALIGN '5s' [TO NOW|ts] [BY (cols)] [FILL v] — RANGE-query clause
        align_ms = align_to = align_by = align_fill = None
        if self.eat_kw("align"):
            align_ms = _duration_ms(str(self.next().value))
            if self.eat_kw("to"):
                if self.eat_kw("now"):
                    align_to = "now"
                elif self.eat_kw("calendar"):
                    align_to = "calendar"
                else:
                    align_to = self.next().value
            if self.eat_kw("by"):
                self.expect_op("(")
                align_by = []
                while not self.eat_op(")"):
                    align_by.append(str(self.next().value))
                    self.eat_op(",")
            if self.eat_kw("fill"):
                align_fill = self._parse_fill()
        order_by = []
        if self.eat_kw("order"):
            self.expect_kw("by")
            while True:
                e = self.parse_expr()
                desc = False
                if self.eat_kw("desc"):
                    desc = True
                else:
                    self.eat_kw("asc")
                order_by.append((e, desc))
                if not self.eat_op(","):
                    break
        limit = offset = None
        if self.eat_kw("limit"):
            limit = int(self.next().value)
        if self.eat_kw("offset"):
            offset = int(self.next().value)
        return ast.Select(projections, table, table_alias=table_alias, joins=joins,
                          where=where, group_by=group_by, having=having,
                          order_by=order_by, limit=limit, offset=offset,
                          align_ms=align_ms, align_to=align_to,
                          align_by=align_by, align_fill=align_fill,
                          distinct=distinct)

    def _parse_fill(self):
        t = self.next()
        if t.kind == "id" and t.value.lower() in ("null", "prev", "linear"):
            return t.value.lower()
        if t.kind == "str" and str(t.value).lower() in ("null", "prev", "linear"):
            return str(t.value).lower()
        neg = False
        if t.kind == "op" and t.value == "-":
            neg, t = True, self.next()
        try:
            v = float(t.value)
        except (TypeError, ValueError):
            raise InvalidSyntax(f"bad FILL value {t.value!r}")
        return -v if neg else v

    def parse_create(self):
        self.expect_kw("create")
        if self.eat_kw("database") or self.eat_kw("schema"):
            if_not_exists = False
            if self.eat_kw("if"):
                self.expect_kw("not"); self.expect_kw("exists")
                if_not_exists = True
            return ast.CreateDatabase(self.next().value, if_not_exists)
        if self.eat_kw("flow"):
            if_not_exists = False
            if self.eat_kw("if"):
                self.expect_kw("not"); self.expect_kw("exists")
                if_not_exists = True
            name = self.next().value
            self.expect_kw("sink")
            self.expect_kw("to")
            sink = self.next().value
            expire_after_s = None
            while not self.at_kw("as") and self.peek() is not None:
                if self.eat_kw("expire"):
                    self.expect_kw("after")
                    t = self.next()
                    if t.kind == "str":
                        expire_after_s = _duration_ms(str(t.value)) // 1000
                    else:
                        expire_after_s = int(t.value)   # seconds
                else:
                    self.next()  # skip COMMENT etc
            self.expect_kw("as")
            if self.peek() is None:
                raise InvalidSyntax("CREATE FLOW ... AS <select> expected")
            sql = self.sql[self.peek().pos:].rstrip().rstrip(";")
            self.i = len(self.toks)
            return ast.CreateFlow(name, sink, sql, if_not_exists,
                                  expire_after_s=expire_after_s)
        if self.at_kw("view") or (self.at_kw("or") and "view" in self.sql.lower()):
            or_replace = False
            if self.eat_kw("or"):
                self.expect_kw("replace")
                or_replace = True
            self.expect_kw("view")
            if_not_exists = False
            if self.eat_kw("if"):
                self.expect_kw("not"); self.expect_kw("exists")
                if_not_exists = True
            name = self.next().value
            self.expect_kw("as")
            if self.peek() is None:
                raise InvalidSyntax("CREATE VIEW ... AS <select> expected")
            sql = self.sql[self.peek().pos:].rstrip().rstrip(";")
            self.parse_query()   # validate the body
            return ast.CreateView(name, sql, or_replace, if_not_exists)
        external = self.eat_kw("external")
        self.expect_kw("table")
        if_not_exists = False
        if self.eat_kw("if"):
            self.expect_kw("not")
            self.expect_kw("exists")
            if_not_exists = True
        name = self.next().value
        if external and not self.at_op("("):
            # schema inferred from the file (reference: file-engine infer)
            options = {}
            while self.eat_kw("with"):
                self.expect_op("(")
                while True:
                    k = self.next().value
                    self.expect_op("=")
                    options[str(k).strip("'")] = self.next().value
                    if not self.eat_op(","):
                        break
                self.expect_op(")")
            return ast.CreateTable(name, [], [], None, if_not_exists, options,
                                   None, external=True)
        self.expect_op("(")
        columns = []
        primary_key: list[str] = []
        time_index = None
        while True:
            if self.at_kw("primary"):
                self.next(); self.expect_kw("key"); self.expect_op("(")
                while True:
                    primary_key.append(self.next().value)
                    if not self.eat_op(","):
                        break
                self.expect_op(")")
            elif self.at_kw("time"):
                self.next(); self.expect_kw("index"); self.expect_op("(")
                time_index = self.next().value
                self.expect_op(")")
            else:
                cname = self.next().value
                ctype = self.next().value
                # optional (precision) e.g. timestamp(3)
                if self.eat_op("("):
                    ctype += f"({self.next().value})"
                    self.expect_op(")")
                opts = {}
                while True:
                    if self.eat_kw("null"):
                        opts["nullable"] = True
                    elif self.eat_kw("not"):
                        self.expect_kw("null")
                        opts["nullable"] = False
                    elif self.eat_kw("time"):
                        self.expect_kw("index")
                        time_index = cname
                    elif self.eat_kw("primary"):
                        self.expect_kw("key")
                        primary_key.append(cname)
                    elif self.eat_kw("default"):
                        opts["default"] = self.parse_expr()
                    elif self.eat_kw("fulltext"):
                        self.eat_kw("index")
                        opts["fulltext"] = True
                        if self.eat_op("("):  # FULLTEXT INDEX WITH-style opts
                            depth = 1
                            while depth and self.peek() is not None:
                                t = self.next()
                                if t.kind == "op" and t.value == "(":
                                    depth += 1
                                elif t.kind == "op" and t.value == ")":
                                    depth -= 1
                    else:
                        break
                columns.append((cname, ctype, opts))
            if not self.eat_op(","):
                break
        self.expect_op(")")
        options = {}
        partitions = None
        partition_on = None
        while self.peek() is not None and not self.at_op(";"):
            if self.eat_kw("with"):
                self.expect_op("(")
                while True:
                    k = self.next().value
                    self.expect_op("=")
                    options[str(k).strip("'")] = self.next().value
                    if not self.eat_op(","):
                        break
                self.expect_op(")")
            elif self.eat_kw("partition"):
                if self.eat_kw("on"):
                    # PARTITION ON COLUMNS (a, b) (expr0, expr1, ...) —
                    # reference src/partition/src/multi_dim.rs; rule engine
                    # in parallel/partition.py
                    self.expect_kw("columns")
                    self.expect_op("(")
                    pcols = []
                    while True:
                        pcols.append(self.next().value)
                        if not self.eat_op(","):
                            break
                    self.expect_op(")")
                    self.expect_op("(")
                    pexprs = []
                    if not self.at_op(")"):
                        while True:
                            pexprs.append(self.parse_expr())
                            if not self.eat_op(","):
                                break
                    self.expect_op(")")
                    partition_on = (pcols, pexprs)
                else:
                    # PARTITION <n> (simplified: region count)
                    partitions = int(self.next().value)
            elif self.eat_kw("engine"):
                self.expect_op("=")
                options["engine"] = self.next().value
            else:
                break
        return ast.CreateTable(name, columns, primary_key, time_index,
                               if_not_exists, options, partitions,
                               external=external, partition_on=partition_on)

    def parse_drop(self):
        self.expect_kw("drop")
        if self.eat_kw("database") or self.eat_kw("schema"):
            if_exists = False
            if self.eat_kw("if"):
                self.expect_kw("exists")
                if_exists = True
            return ast.DropDatabase(self.next().value, if_exists)
        if self.eat_kw("flow"):
            return ast.DropFlow(self.next().value)
        if self.eat_kw("view"):
            if_exists = False
            if self.eat_kw("if"):
                self.expect_kw("exists")
                if_exists = True
            return ast.DropView(self.next().value, if_exists)
        self.expect_kw("table")
        if_exists = False
        if self.eat_kw("if"):
            self.expect_kw("exists")
            if_exists = True
        return ast.DropTable(self.next().value, if_exists)

    def parse_insert(self) -> ast.InsertValues:
        self.expect_kw("insert")
        self.expect_kw("into")
        table = self.next().value
        columns = []
        if self.eat_op("("):
            while True:
                columns.append(self.next().value)
                if not self.eat_op(","):
                    break
            self.expect_op(")")
        if self.at_kw("select") or self.at_kw("with"):
            return ast.InsertValues(table, columns, [],
                                    select=self.parse_query())
        self.expect_kw("values")
        rows = []
        while True:
            self.expect_op("(")
            row = []
            while True:
                e = self.parse_expr()
                if isinstance(e, ast.Lit):
                    row.append(e.value)
                elif isinstance(e, ast.UnaryOp) and e.op == "-" and isinstance(e.operand, ast.Lit):
                    row.append(-e.operand.value)
                else:
                    # constant expressions (now(), 1+2, CAST(...)) fold at
                    # parse time (reference: sqlparser values exprs)
                    from greptimedb_amd.query.executor import _eval_const
                    try:
                        row.append(_eval_const(e))
                    except Exception:
                        raise InvalidSyntax(
                            "INSERT VALUES must be constant expressions") \
                            from None
                if not self.eat_op(","):
                    break
            self.expect_op(")")
            rows.append(row)
            if not self.eat_op(","):
                break
        return ast.InsertValues(table, columns, rows)

    def parse_tql(self) -> ast.Tql:
        self.expect_kw("tql")
        self.expect_kw("eval")
        self.expect_op("(")
        start = float(self.next().value)
        self.expect_op(",")
        end = float(self.next().value)
        self.expect_op(",")
        step_tok = self.next()
        step = parse_interval_text(step_tok.value) / 1000 if isinstance(step_tok.value, str) \
            else float(step_tok.value)
        self.expect_op(")")
        # the rest of the ORIGINAL string is the raw PromQL text (the SQL
        # lexer cannot tokenize PromQL selectors)
        if self.peek() is not None:
            query = self.sql[self.peek().pos:]
        else:
            query = ""
        query = query.rstrip().rstrip(";")
        self.i = len(self.toks)  # consume everything
        return ast.Tql(start, end, step, query)

    # ---------------- expressions (Pratt) ----------------
    def _parse_values_tail(self) -> "ast.ValuesTable":
        """VALUES (a, b), (c, d) — literal row constructor
        (reference: sqlparser VALUES table factor)."""
        self.expect_kw("values")
        rows = []
        while True:
            self.expect_op("(")
            row = []
            while True:
                e = self.parse_expr()
                from greptimedb_amd.query.executor import _eval_const
                row.append(_eval_const(e))
                if not self.eat_op(","):
                    break
            self.expect_op(")")
            rows.append(row)
            if not self.eat_op(","):
                break
        return ast.ValuesTable(rows)

    def _parse_type_name(self) -> str:
        """Type name after CAST(... AS …) / `::` — single identifier with
        optional (n[,m]) length args (ignored) and the DOUBLE PRECISION /
        TIMESTAMP(3) spellings."""
        t = self.next()
        name = str(t.value).lower()
        if name == "double" and self.eat_kw("precision"):
            pass
        if self.eat_op("("):
            while not self.eat_op(")"):
                self.next()
        return name

    def parse_expr(self, min_prec: int = 0) -> ast.Expr:
        left = self.parse_prefix()
        while self.eat_op("::"):       # postgres cast: expr::TYPE
            left = ast.Cast(left, self._parse_type_name())
        while True:
            t = self.peek()
            if t is None:
                break
            opname = None
            negated = False
            if t.kind == "op" and t.value in _PRECEDENCE:
                opname = t.value
            elif t.kind == "id" and t.value.lower() in _PRECEDENCE:
                opname = t.value.lower()
            elif t.kind == "id" and t.value.lower() == "not" and \
                    self.i + 1 < len(self.toks) and \
                    self.toks[self.i + 1].kind == "id" and \
                    self.toks[self.i + 1].value.lower() in ("in", "like",
                                                            "between"):
                self.next()                       # consume NOT
                opname = self.toks[self.i].value.lower()
                negated = True
            if opname is None:
                break
            prec = _PRECEDENCE[opname]
            if prec <= min_prec:
                break
            # special postfix-ish operators
            if opname == "in":
                self.next()
                self.expect_op("(")
                if self.at_kw("select"):   # IN (SELECT ...) subquery
                    sub = self.parse_select()
                    self.expect_op(")")
                    left = ast.InList(left, [ast.ScalarSubquery(sub, many=True)],
                                      negated=negated)
                    continue
                items = []
                while True:
                    items.append(self.parse_expr())
                    if not self.eat_op(","):
                        break
                self.expect_op(")")
                left = ast.InList(left, items, negated=negated)
                continue
            if opname == "between":
                self.next()
                low = self.parse_expr(_PRECEDENCE["between"])
                self.expect_kw("and")
                high = self.parse_expr(_PRECEDENCE["between"])
                left = ast.Between(left, low, high, negated=negated)
                continue
            if opname == "is":
                self.next()
                negated = self.eat_kw("not")
                self.expect_kw("null")
                left = ast.IsNull(left, negated)
                continue
            self.next()
            if t.kind == "id" and opname == "like":
                right = self.parse_expr(prec)
                left = ast.BinOp("like", left, right)
                if negated:
                    left = ast.UnaryOp("not", left)
                continue
            right = self.parse_expr(prec)
            left = ast.BinOp(opname, left, right)
        return left

    def parse_prefix(self) -> ast.Expr:
        t = self.next()
        if t.kind == "num":
            return ast.Lit(t.value)
        if t.kind == "str":
            return ast.Lit(t.value)
        if t.kind == "op" and t.value == "(":
            if self.at_kw("select"):   # scalar subquery (uncorrelated)
                sub = self.parse_select()
                self.expect_op(")")
                return ast.ScalarSubquery(sub)
            if self.at_kw("values"):   # VALUES row constructor
                return self._parse_values_tail()
            e = self.parse_expr()
            self.expect_op(")")
            return e
        if t.kind == "op" and t.value == "-":
            return ast.UnaryOp("-", self.parse_expr(7))
        if t.kind == "op" and t.value == "@" and self.eat_op("@"):
            # @@name / @@session.name system variables (MySQL dialect)
            return ast.SysVar(str(self.next().value).lower())
        if t.kind == "op" and t.value == "*":
            return ast.Star()
        if t.kind == "id":
            low = t.value.lower()
            if low == "not":
                return ast.UnaryOp("not", self.parse_expr(3))
            if low == "exists" and self.at_op("("):
                self.next()
                sub = self.parse_select()
                self.expect_op(")")
                return ast.Exists(sub)
            if low == "cast" and self.at_op("("):
                self.next()
                inner = self.parse_expr()
                self.expect_kw("as")
                ty = self._parse_type_name()
                self.expect_op(")")
                return ast.Cast(inner, ty)
            if low == "interval":
                txt = self.next().value
                return ast.Interval(parse_interval_text(str(txt)), str(txt))
            if low == "case":
                operand = None if self.at_kw("when") else self.parse_expr()
                whens = []
                while self.eat_kw("when"):
                    w = self.parse_expr()
                    self.expect_kw("then")
                    whens.append((w, self.parse_expr()))
                default = self.parse_expr() if self.eat_kw("else") else None
                self.expect_kw("end")
                return ast.Case(operand, whens, default)
            if low in ("true", "false"):
                return ast.Lit(low == "true")
            if low == "null":
                return ast.Lit(None)
            if self.at_op("("):
                self.next()
                args = []
                distinct = False
                if self.eat_kw("distinct"):
                    distinct = True
                if not self.at_op(")"):
                    while True:
                        if self.eat_op("*"):
                            args.append(ast.Star())
                        else:
                            args.append(self.parse_expr())
                        if not self.eat_op(","):
                            break
                # last_value(v ORDER BY ts [DESC]) / first_value(...) — the
                # reference's ordered-set form; DESC flips first↔last
                if low in ("last_value", "first_value") and self.eat_kw("order"):
                    self.expect_kw("by")
                    self.next()  # the order column (time index)
                    if self.eat_kw("desc"):
                        low = ("first_value" if low == "last_value"
                               else "last_value")
                    else:
                        self.eat_kw("asc")
                self.expect_op(")")
                if self.eat_kw("over"):
                    return self._window_spec(low, args)
                fn = ast.Func(low, args, distinct)
                # agg(col) RANGE '10s' [FILL v] binds tighter than operators
                if self.at_kw("range") and self.i + 1 < len(self.toks) and \
                        self.toks[self.i + 1].kind == "str":
                    self.next()
                    dur = str(self.next().value)
                    fill = self._parse_fill() if self.eat_kw("fill") else None
                    return ast.RangeAgg(fn, _duration_ms(dur), fill)
                return fn
            return ast.Col(t.value)
        raise InvalidSyntax(f"unexpected token {t}")

    def _window_spec(self, func: str, args: list) -> "ast.WindowFunc":
        """OVER (PARTITION BY … ORDER BY … [ROWS|RANGE frame]) — frames are
        accepted only in their default forms (ref: DataFusion WindowExpr)."""
        self.expect_op("(")
        part, order = [], []
        if self.eat_kw("partition"):
            self.expect_kw("by")
            while True:
                part.append(self.parse_expr())
                if not self.eat_op(","):
                    break
        if self.eat_kw("order"):
            self.expect_kw("by")
            while True:
                e2 = self.parse_expr()
                desc = False
                if self.eat_kw("desc"):
                    desc = True
                else:
                    self.eat_kw("asc")
                order.append((e2, desc))
                if not self.eat_op(","):
                    break
        if self.at_kw("rows", "range"):
            # accept the frames equivalent to the defaults:
            #   ROWS/RANGE BETWEEN UNBOUNDED PRECEDING AND CURRENT ROW
            self.next()
            self.expect_kw("between")
            self.expect_kw("unbounded")
            self.expect_kw("preceding")
            self.expect_kw("and")
            self.expect_kw("current")
            self.expect_kw("row")
        self.expect_op(")")
        return ast.WindowFunc(func, args, part, order)


def parse_sql(sql: str):
    return Parser(sql).parse_statement()
