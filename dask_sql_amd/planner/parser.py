"""Minimal SQL frontend (SELECT subset) → AST.

Replaces the *interface* of the reference's Rust planner entry points
(src/sql.rs:570 parse_sql, :586 logical_relational_algebra) for the hot-path
query shapes: SELECT-project-filter-join-groupby-
orderby-limit with the rex op subset of SURVEY §2. Plan semantics are pinned
end-to-end by the golden tests, not structurally.
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field


# ---- AST ------------------------------------------------------------------
@dataclass
class TableRef:
    name: str | None
    alias: str | None = None
    subquery: object = None  # SelectStmt for derived tables


@dataclass
class JoinClause:
    join_type: str  # INNER/LEFT/RIGHT/FULL/LEFTSEMI/LEFTANTI/CROSS
    table: TableRef
    on: tuple | None


@dataclass
class SelectStmt:
    items: list  # [(expr_ast, alias|None)] ; expr_ast ('star',) allowed
    # (UnionStmt defined below wraps several SelectStmts)
    distinct: bool = False
    from_tables: list = field(default_factory=list)  # [TableRef]
    joins: list = field(default_factory=list)  # [JoinClause]
    where: tuple | None = None
    group_by: list = field(default_factory=list)
    having: tuple | None = None
    order_by: list = field(default_factory=list)  # [(expr, asc, nulls_first)]
    limit: int | None = None
    offset: int = 0


@dataclass
class UnionStmt:
    branches: list  # [SelectStmt]
    alls: list      # [bool] per UNION step (False = UNION DISTINCT)
    order_by: list = field(default_factory=list)
    limit: int | None = None
    offset: int = 0


KEYWORDS = {
    "SELECT", "DISTINCT", "FROM", "WHERE", "GROUP", "BY", "HAVING", "ORDER",
    "LIMIT", "OFFSET", "AS", "AND", "OR", "NOT", "JOIN", "INNER", "LEFT",
    "RIGHT", "FULL", "OUTER", "SEMI", "ANTI", "CROSS", "ON", "TRUE", "FALSE",
    "NULL", "IS", "IN", "BETWEEN", "LIKE", "CASE", "WHEN", "THEN", "ELSE",
    "END", "CAST", "DATE", "ASC", "DESC", "NULLS", "FIRST", "LAST", "FILTER",
    "TIMESTAMP", "INTERVAL", "UNION", "ALL", "OVER", "PARTITION", "EXISTS",
}

WINDOW_FUNCS = {"ROW_NUMBER", "RANK", "DENSE_RANK", "SUM", "COUNT", "AVG",
                "MIN", "MAX", "LAG", "LEAD", "FIRST_VALUE"}

_TOKEN_RE = re.compile(
    r"""
    (?P<ws>\s+)
  | (?P<num>\d+\.\d*(?:[eE][+-]?\d+)?|\.\d+(?:[eE][+-]?\d+)?|\d+(?:[eE][+-]?\d+)?)
  | (?P<str>'(?:[^']|'')*')
  | (?P<qid>"[^"]+")
  | (?P<id>[A-Za-z_][A-Za-z_0-9]*)
  | (?P<op>\|\||<>|!=|>=|<=|=|<|>|\+|-|\*|/|%|\(|\)|,|\.)
    """,
    re.VERBOSE,
)


def tokenize(sql: str):
    toks = []
    pos = 0
    while pos < len(sql):
        m = _TOKEN_RE.match(sql, pos)
        if not m:
            raise ValueError(f"SQL tokenize error at: {sql[pos:pos+30]!r}")
        pos = m.end()
        if m.lastgroup == "ws":
            continue
        text = m.group()
        if m.lastgroup == "id" and text.upper() in KEYWORDS:
            toks.append(("kw", text.upper()))
        elif m.lastgroup == "id":
            toks.append(("id", text))
        elif m.lastgroup == "qid":
            toks.append(("id", text[1:-1]))
        elif m.lastgroup == "num":
            toks.append(("num", text))
        elif m.lastgroup == "str":
            toks.append(("str", text[1:-1].replace("''", "'")))
        else:
            toks.append(("op", text))
    toks.append(("eof", ""))
    return toks


AGG_FUNCS = {"SUM", "COUNT", "AVG", "MIN", "MAX", "ANY_VALUE", "STDDEV",
             "STDDEV_POP", "STDDEV_SAMP", "VAR_SAMP", "VAR_POP", "VARIANCE",
             "SINGLE_VALUE", "EVERY", "BOOL_AND", "BOOL_OR"}


class Parser:
    def __init__(self, sql: str):
        self.toks = tokenize(sql)
        self.i = 0

    # -- token helpers ------------------------------------------------------
    def peek(self):
        return self.toks[self.i]

    def next(self):
        t = self.toks[self.i]
        self.i += 1
        return t

    def accept_kw(self, *kws):
        t = self.peek()
        if t[0] == "kw" and t[1] in kws:
            self.next()
            return t[1]
        return None

    def expect_kw(self, kw):
        t = self.next()
        if t != ("kw", kw):
            raise ValueError(f"expected {kw}, got {t}")

    def accept_op(self, *ops):
        t = self.peek()
        if t[0] == "op" and t[1] in ops:
            self.next()
            return t[1]
        return None

    def expect_op(self, op):
        t = self.next()
        if t != ("op", op):
            raise ValueError(f"expected {op!r}, got {t}")

    # -- entry --------------------------------------------------------------
    def parse(self):
        stmt = self.select_stmt()
        branches = [stmt]
        alls = []
        while self.accept_kw("UNION"):
            alls.append(bool(self.accept_kw("ALL")))
            branches.append(self.select_stmt())
        if self.peek()[0] != "eof":
            raise ValueError(f"trailing tokens: {self.peek()}")
        if len(branches) == 1:
            return stmt
        # trailing ORDER BY / LIMIT bind to the whole union, not the last
        # branch (standard SQL)
        last = branches[-1]
        u = UnionStmt(branches=branches, alls=alls,
                      order_by=last.order_by, limit=last.limit,
                      offset=last.offset)
        last.order_by, last.limit, last.offset = [], None, 0
        return u

    def select_stmt(self) -> SelectStmt:
        self.expect_kw("SELECT")
        s = SelectStmt(items=[])
        if self.accept_kw("DISTINCT"):
            s.distinct = True
        # select list
        while True:
            if self.accept_op("*"):
                s.items.append((("star",), None))
            else:
                e = self.expr()
                alias = None
                if self.accept_kw("AS"):
                    alias = self._name()
                elif self.peek()[0] == "id":
                    alias = self._name()
                s.items.append((e, alias))
            if not self.accept_op(","):
                break
        if self.accept_kw("FROM"):
            s.from_tables.append(self.table_ref())
            while True:
                if self.accept_op(","):
                    s.from_tables.append(self.table_ref())
                    continue
                jt = self._join_type()
                if jt is None:
                    break
                tr = self.table_ref()
                on = None
                if self.accept_kw("ON"):
                    on = self.expr()
                s.joins.append(JoinClause(jt, tr, on))
        if self.accept_kw("WHERE"):
            s.where = self.expr()
        if self.accept_kw("GROUP"):
            self.expect_kw("BY")
            s.group_by.append(self.expr())
            while self.accept_op(","):
                s.group_by.append(self.expr())
        if self.accept_kw("HAVING"):
            s.having = self.expr()
        if self.accept_kw("ORDER"):
            self.expect_kw("BY")
            while True:
                e = self.expr()
                asc = True
                if self.accept_kw("DESC"):
                    asc = False
                else:
                    self.accept_kw("ASC")
                nulls_first = not asc  # SQL default: NULLS LAST for ASC
                if self.accept_kw("NULLS"):
                    nulls_first = self.accept_kw("FIRST") is not None
                    if not nulls_first:
                        self.expect_kw("LAST")
                s.order_by.append((e, asc, nulls_first))
                if not self.accept_op(","):
                    break
        if self.accept_kw("LIMIT"):
            s.limit = int(self.next()[1])
        if self.accept_kw("OFFSET"):
            s.offset = int(self.next()[1])
        return s

    def _join_type(self):
        if self.accept_kw("JOIN"):
            return "INNER"
        if self.accept_kw("INNER"):
            self.expect_kw("JOIN")
            return "INNER"
        if self.accept_kw("LEFT"):
            if self.accept_kw("OUTER"):
                self.expect_kw("JOIN")
                return "LEFT"
            if self.accept_kw("SEMI"):
                self.expect_kw("JOIN")
                return "LEFTSEMI"
            if self.accept_kw("ANTI"):
                self.expect_kw("JOIN")
                return "LEFTANTI"
            self.expect_kw("JOIN")
            return "LEFT"
        if self.accept_kw("RIGHT"):
            self.accept_kw("OUTER")
            self.expect_kw("JOIN")
            return "RIGHT"
        if self.accept_kw("FULL"):
            self.accept_kw("OUTER")
            self.expect_kw("JOIN")
            return "FULL"
        if self.accept_kw("CROSS"):
            self.expect_kw("JOIN")
            return "CROSS"
        return None

    def table_ref(self) -> TableRef:
        if self.peek() == ("op", "("):
            # derived table: FROM (SELECT ...) [AS] alias
            self.next()
            sub = self.select_stmt()
            branches, alls = [sub], []
            while self.accept_kw("UNION"):
                alls.append(bool(self.accept_kw("ALL")))
                branches.append(self.select_stmt())
            if len(branches) > 1:
                sub = UnionStmt(branches=branches, alls=alls)
            self.expect_op(")")
            alias = None
            if self.accept_kw("AS"):
                alias = self._name()
            elif self.peek()[0] == "id":
                alias = self._name()
            if alias is None:
                raise ValueError("derived table needs an alias")
            return TableRef(name=None, alias=alias, subquery=sub)
        name = self._name()
        alias = None
        if self.accept_kw("AS"):
            alias = self._name()
        elif self.peek()[0] == "id":
            alias = self._name()
        return TableRef(name, alias)

    def _name(self) -> str:
        t = self.next()
        if t[0] not in ("id",):
            raise ValueError(f"expected identifier, got {t}")
        return t[1]

    # -- expressions ---------------------------------------------------------
    def expr(self):
        return self.or_expr()

    def or_expr(self):
        e = self.and_expr()
        while self.accept_kw("OR"):
            e = ("call", "OR", [e, self.and_expr()])
        return e

    def and_expr(self):
        e = self.not_expr()
        while self.accept_kw("AND"):
            e = ("call", "AND", [e, self.not_expr()])
        return e

    def not_expr(self):
        if self.accept_kw("NOT"):
            return ("call", "NOT", [self.not_expr()])
        return self.cmp_expr()

    def cmp_expr(self):
        e = self.add_expr()
        while True:
            op = self.accept_op("=", "<>", "!=", "<", "<=", ">", ">=")
            if op:
                rhs = self.add_expr()
                e = ("call", "<>" if op == "!=" else op, [e, rhs])
                continue
            if self.accept_kw("IS"):
                neg = self.accept_kw("NOT") is not None
                self.expect_kw("NULL")
                e = ("call", "IS NOT NULL" if neg else "IS NULL", [e])
                continue
            if self.accept_kw("BETWEEN"):
                lo = self.add_expr()
                self.expect_kw("AND")
                hi = self.add_expr()
                e = ("call", "AND",
                     [("call", ">=", [e, lo]), ("call", "<=", [e, hi])])
                continue
            if self.accept_kw("LIKE"):
                pat = self.add_expr()
                e = ("call", "LIKE", [e, pat])
                continue
            if self.accept_kw("NOT"):
                if self.accept_kw("IN"):
                    e = ("call", "NOT", [self._in_list(e)])
                    continue
                if self.accept_kw("LIKE"):
                    pat = self.add_expr()
                    e = ("call", "NOT", [("call", "LIKE", [e, pat])])
                    continue
                if self.accept_kw("BETWEEN"):
                    lo = self.add_expr()
                    self.expect_kw("AND")
                    hi = self.add_expr()
                    e = ("call", "NOT",
                         [("call", "AND",
                           [("call", ">=", [e, lo]), ("call", "<=", [e, hi])])])
                    continue
                raise ValueError("expected IN or BETWEEN after NOT")
            if self.accept_kw("IN"):
                e = self._in_list(e)
                continue
            break
        return e

    def _in_list(self, e):
        self.expect_op("(")
        if self.peek() == ("kw", "SELECT"):
            # IN (SELECT ...) — decorrelated to a SEMI/ANTI join by the
            # builder (what DataFusion's subquery rewriting gives the
            # reference)
            sub = self.select_stmt()
            self.expect_op(")")
            return ("in_sub", e, sub)
        items = [self.expr()]
        while self.accept_op(","):
            items.append(self.expr())
        self.expect_op(")")
        out = ("call", "=", [e, items[0]])
        for it in items[1:]:
            out = ("call", "OR", [out, ("call", "=", [e, it])])
        return out

    def add_expr(self):
        e = self.mul_expr()
        while True:
            op = self.accept_op("+", "-", "||")
            if not op:
                break
            rhs = self.mul_expr()
            if op == "||":
                e = ("call", "CONCAT", [e, rhs])
            else:
                e = ("call", op, [e, rhs])
        return e

    def mul_expr(self):
        e = self.unary_expr()
        while True:
            op = self.accept_op("*", "/", "%")
            if not op:
                break
            if op == "%":
                e = ("call", "MOD", [e, self.unary_expr()])
            else:
                e = ("call", op, [e, self.unary_expr()])
        return e

    def unary_expr(self):
        if self.accept_op("-"):
            return ("call", "NEG", [self.unary_expr()])
        self.accept_op("+")
        return self.primary()

    def primary(self):
        t = self.peek()
        if t == ("op", "("):
            self.next()
            if self.peek() == ("kw", "SELECT"):
                sub = self.select_stmt()
                self.expect_op(")")
                return ("scalar_sub", sub)
            e = self.expr()
            self.expect_op(")")
            return e
        if t[0] == "num":
            self.next()
            txt = t[1]
            if "." in txt or "e" in txt.lower():
                return ("lit", float(txt), "DOUBLE")
            return ("lit", int(txt), "BIGINT")
        if t[0] == "str":
            self.next()
            return ("lit", t[1], "VARCHAR")
        if t[0] == "kw":
            if t[1] in ("TRUE", "FALSE"):
                self.next()
                return ("lit", t[1] == "TRUE", "BOOLEAN")
            if t[1] == "NULL":
                self.next()
                return ("lit", None, "NULL")
            if t[1] == "DATE":
                self.next()
                s = self.next()
                if s[0] != "str":
                    raise ValueError("DATE needs a string literal")
                return ("lit", s[1], "DATE")
            if t[1] == "TIMESTAMP":
                self.next()
                s = self.next()
                if s[0] != "str":
                    raise ValueError("TIMESTAMP needs a string literal")
                return ("lit", s[1], "TIMESTAMP")
            if t[1] == "INTERVAL":
                # INTERVAL '<n>' DAY|WEEK|MONTH|YEAR (TPC-H date arithmetic)
                self.next()
                s = self.next()
                if s[0] not in ("str", "num"):
                    raise ValueError("INTERVAL needs a quantity literal")
                u = self.next()
                unit = str(u[1]).upper().rstrip("S")
                if unit not in ("DAY", "WEEK", "MONTH", "YEAR"):
                    raise ValueError(f"INTERVAL unit {u[1]!r} not supported")
                return ("interval", int(str(s[1]).strip()), unit)
            if t[1] == "EXISTS":
                self.next()
                self.expect_op("(")
                sub = self.select_stmt()
                self.expect_op(")")
                return ("exists", sub)
            if t[1] == "CASE":
                return self._case()
            if t[1] == "CAST":
                self.next()
                self.expect_op("(")
                e = self.expr()
                self.expect_kw("AS")
                ty = self._type_name()
                self.expect_op(")")
                return ("cast", e, ty)
        if t[0] == "id":
            name = self._name()
            if name.upper() == "TRIM" and self.peek() == ("op", "("):
                # TRIM([LEADING|TRAILING|BOTH] ['ch'] FROM x) | TRIM(x)
                self.next()
                mode = "BOTH"
                if self.peek()[0] == "id" and self.peek()[1].upper() in (
                        "LEADING", "TRAILING", "BOTH"):
                    mode = self.next()[1].upper()
                ch = " "
                if self.peek()[0] == "str":
                    ch = self.next()[1]
                if self.accept_kw("FROM"):
                    e = self.expr()
                else:
                    e = self.expr()
                self.expect_op(")")
                return ("call", "TRIM",
                        [e, ("lit", mode, "VARCHAR"),
                         ("lit", ch, "VARCHAR")])
            if name.upper() in ("SUBSTRING", "SUBSTR") \
                    and self.peek() == ("op", "("):
                # SUBSTRING(x FROM a [FOR n]) — Calcite form
                save = self.i
                self.next()
                e = self.expr()
                if self.accept_kw("FROM"):
                    start = self.expr()
                    args = [e, start]
                    if self.peek() == ("id", "FOR") or \
                            self.accept_kw("FOR") or \
                            (self.peek()[0] == "id"
                             and self.peek()[1].upper() == "FOR"):
                        if self.peek()[0] == "id":
                            self.next()
                        args.append(self.expr())
                    self.expect_op(")")
                    return ("call", "SUBSTRING", args)
                self.i = save  # comma form: reparse generically
            if name.upper() == "EXTRACT" and self.peek() == ("op", "("):
                # EXTRACT(field FROM expr)
                self.next()
                ft = self.next()
                field = ft[1].upper()
                if field not in ("YEAR", "MONTH", "DAY", "HOUR", "MINUTE",
                                 "SECOND"):
                    raise ValueError(f"EXTRACT({field}) not supported")
                self.expect_kw("FROM")
                e = self.expr()
                self.expect_op(")")
                return ("call", f"EXTRACT_{field}", [e])
            if self.peek() == ("op", "("):
                return self._func_call(name)
            if self.accept_op("."):
                col = self._name()
                return ("col", name, col)
            return ("col", None, name)
        raise ValueError(f"unexpected token {t}")

    def _case(self):
        self.expect_kw("CASE")
        # simple CASE (CASE x WHEN v THEN ...) rewrites to searched form
        operand = None
        if self.peek() != ("kw", "WHEN"):
            operand = self.expr()
        whens = []
        while self.accept_kw("WHEN"):
            cond = self.expr()
            if operand is not None:
                cond = ("call", "=", [operand, cond])
            self.expect_kw("THEN")
            val = self.expr()
            whens.append((cond, val))
        els = None
        if self.accept_kw("ELSE"):
            els = self.expr()
        self.expect_kw("END")
        return ("case", whens, els)

    def _type_name(self):
        t = self.next()
        name = t[1].upper()
        # swallow (p[,s])
        if self.accept_op("("):
            while not self.accept_op(")"):
                self.next()
        return name

    def _func_call(self, name):
        fname = name.upper()
        self.expect_op("(")
        distinct = False
        args = []
        if self.accept_op("*"):
            args = [("star",)]
        elif self.peek() != ("op", ")"):
            if self.accept_kw("DISTINCT"):
                distinct = True
            args.append(self.expr())
            while self.accept_op(","):
                args.append(self.expr())
        self.expect_op(")")
        filter_expr = None
        if self.peek() == ("kw", "FILTER"):
            self.next()
            self.expect_op("(")
            self.expect_kw("WHERE")
            filter_expr = self.expr()
            self.expect_op(")")
        if self.peek() == ("kw", "OVER"):
            if fname not in WINDOW_FUNCS:
                raise ValueError(f"{fname} is not a supported window function")
            self.next()
            self.expect_op("(")
            part, order = [], []
            if self.accept_kw("PARTITION"):
                self.expect_kw("BY")
                part.append(self.expr())
                while self.accept_op(","):
                    part.append(self.expr())
            if self.accept_kw("ORDER"):
                self.expect_kw("BY")
                while True:
                    e = self.expr()
                    desc = bool(self.accept_kw("DESC"))
                    if not desc:
                        self.accept_kw("ASC")
                    order.append((e, desc))
                    if not self.accept_op(","):
                        break
            self.expect_op(")")
            return ("window", fname.lower(), args, tuple(part), tuple(order))
        if fname in AGG_FUNCS:
            return ("agg", fname.lower(), args, distinct, filter_expr)
        return ("call", fname, args)


def parse_sql(sql: str) -> SelectStmt:
    return Parser(sql.strip().rstrip(";")).parse()
