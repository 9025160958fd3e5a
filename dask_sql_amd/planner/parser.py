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
    ops: list | None = None  # per step: UNION | INTERSECT | EXCEPT


KEYWORDS = {
    "SELECT", "DISTINCT", "FROM", "WHERE", "GROUP", "BY", "HAVING", "ORDER",
    "LIMIT", "OFFSET", "AS", "AND", "OR", "NOT", "JOIN", "INNER", "LEFT",
    "RIGHT", "FULL", "OUTER", "SEMI", "ANTI", "CROSS", "ON", "TRUE", "FALSE",
    "NULL", "IS", "IN", "BETWEEN", "LIKE", "CASE", "WHEN", "THEN", "ELSE",
    "END", "CAST", "DATE", "ASC", "DESC", "NULLS", "FIRST", "LAST", "FILTER",
    "TIMESTAMP", "INTERVAL", "UNION", "ALL", "OVER", "PARTITION", "EXISTS",
}

WINDOW_FUNCS = {"ROW_NUMBER", "RANK", "DENSE_RANK", "SUM", "COUNT", "AVG",
                "MIN", "MAX", "LAG", "LEAD", "FIRST_VALUE", "LAST_VALUE",
                "SINGLE_VALUE"}

_TOKEN_RE = re.compile(
    r"""
    (?P<ws>\s+|--[^\n]*|/\*[\s\S]*?\*/)
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


# words that terminate an implicit-alias position (reserved in standard SQL
# but lexed as identifiers here)
_NON_ALIAS = {"INTERSECT", "EXCEPT", "DISTRIBUTE"}

AGG_FUNCS = {"SUM", "COUNT", "AVG", "MIN", "MAX", "ANY_VALUE", "STDDEV",
             "STDDEV_POP", "STDDEV_SAMP", "VAR_SAMP", "VAR_POP", "VARIANCE",
             "SINGLE_VALUE", "EVERY", "BOOL_AND", "BOOL_OR",
             "BIT_AND", "BIT_OR", "BIT_XOR"}


class Parser:
    def __init__(self, sql: str):
        self.toks = tokenize(sql)
        self.i = 0
        self.ctes = {}  # lowercase name -> SelectStmt/UnionStmt (WITH ...)

    # -- token helpers ------------------------------------------------------
    def peek(self):
        return self.toks[min(self.i, len(self.toks) - 1)]

    def next(self):
        if self.i >= len(self.toks):
            raise ValueError("unexpected end of SQL")
        t = self.toks[self.i]
        self.i += 1
        return t

    # peek() never walks past the trailing ("eof", "") sentinel


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

    def _accept_word(self, w):
        """Accept a non-reserved word (ILIKE/SIMILAR/TO/ESCAPE ...) given as
        an identifier or keyword token, case-insensitively."""
        t = self.peek()
        if t[0] in ("id", "kw") and str(t[1]).upper() == w:
            self.next()
            return True
        return False

    def _expect_word(self, w):
        if not self._accept_word(w):
            raise ValueError(f"expected {w}, got {self.peek()}")

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
        # WITH name AS (select) [, name2 AS (...)] <stmt> — CTEs register as
        # named derived tables; every FROM reference inlines a deep copy of
        # the definition (the reference hands CTEs to DataFusion, which
        # inlines them the same way for non-recursive WITH).
        if self._accept_word("WITH"):
            if self._accept_word("RECURSIVE"):
                raise ValueError("WITH RECURSIVE not supported")
            while True:
                cname = self._name()
                self.expect_kw("AS")
                self.expect_op("(")
                sub = self._set_tail(self.select_stmt())
                self.expect_op(")")
                self.ctes[cname.lower()] = sub
                if not self.accept_op(","):
                    break
        stmt = self._set_tail(self._select_branch())
        if self.peek()[0] != "eof":
            raise ValueError(f"trailing tokens: {self.peek()}")
        if not isinstance(stmt, UnionStmt):
            return stmt
        # trailing ORDER BY / LIMIT bind to the whole set expression, not
        # the last branch (standard SQL)
        last = stmt.branches[-1]
        while isinstance(last, UnionStmt):
            last = last.branches[-1]
        stmt.order_by, stmt.limit, stmt.offset = (last.order_by, last.limit,
                                                  last.offset)
        last.order_by, last.limit, last.offset = [], None, 0
        return stmt

    def _select_branch(self):
        """One branch of a set expression: SELECT ... or a parenthesized
        select / set expression."""
        if self.peek() == ("op", "("):
            save = self.i
            self.next()
            if self.peek() == ("kw", "SELECT") or self.peek() == ("op",
                                                                  "("):
                sub = self._set_tail(self._select_branch())
                self.expect_op(")")
                return sub
            self.i = save  # a parenthesized expression, not a select
        return self.select_stmt()

    def _set_tail(self, first):
        """UNION [ALL] / INTERSECT / EXCEPT chain after a select;
        INTERSECT binds tighter than UNION/EXCEPT (standard SQL)."""
        def chain(left):
            while self._accept_word("INTERSECT"):
                if self.accept_kw("ALL"):
                    raise ValueError("INTERSECT ALL not supported")
                left = UnionStmt([left, self._select_branch()],
                                 alls=[False], ops=["INTERSECT"])
            return left

        stmt = chain(first)
        branches, alls, ops = [stmt], [], []
        while True:
            if self.accept_kw("UNION"):
                op, allf = "UNION", bool(self.accept_kw("ALL"))
            elif self._accept_word("EXCEPT"):
                if self.accept_kw("ALL"):
                    raise ValueError("EXCEPT ALL not supported")
                op, allf = "EXCEPT", False
            else:
                break
            branches.append(chain(self._select_branch()))
            alls.append(allf)
            ops.append(op)
        if len(branches) == 1:
            return stmt
        return UnionStmt(branches, alls, ops=ops)

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
                    # after an explicit AS any word may serve as the alias,
                    # including reserved ones (`... AS date`)
                    t = self.next()
                    if t[0] not in ("id", "kw"):
                        raise ValueError(f"expected alias after AS, got {t}")
                    alias = t[1] if t[0] == "id" else t[1].lower()
                elif self.peek()[0] == "id" \
                        and self.peek()[1].upper() not in _NON_ALIAS:
                    alias = self._name()
                s.items.append((e, alias))
            if not self.accept_op(","):
                break
            if self.peek() == ("kw", "FROM"):
                break  # tolerated trailing comma (reference test_rex.py)
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
        if self._accept_word("DISTRIBUTE"):
            # DISTRIBUTE BY <cols>: a partitioning hint in the reference
            # (rel/custom/distributeby.py → dask shuffle). Row content is
            # unchanged; in one process the shuffle is the identity, and the
            # RCCL exchange path partitions explicitly (distributed.py) —
            # parse and accept, keep the plan unchanged.
            self.expect_kw("BY")
            self.expr()
            while self.accept_op(","):
                self.expr()
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
            if self.accept_kw("ALL"):
                pass  # LIMIT ALL = unlimited
            else:
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
            sub = self._set_tail(self.select_stmt())
            self.expect_op(")")
            alias = None
            if self.accept_kw("AS"):
                alias = self._name()
            elif self.peek()[0] == "id" \
                    and self.peek()[1].upper() not in _NON_ALIAS:
                alias = self._name()
            if alias is None:
                self._dt_n = getattr(self, "_dt_n", 0) + 1
                alias = f"_dt{self._dt_n}"
            return TableRef(name=None, alias=alias, subquery=sub)
        name = self._name()
        alias = None
        if self.accept_kw("AS"):
            alias = self._name()
        elif self.peek()[0] == "id" \
                and self.peek()[1].upper() not in _NON_ALIAS:
            alias = self._name()
        if name.lower() in self.ctes:
            import copy
            return TableRef(name=None, alias=alias or name,
                            subquery=copy.deepcopy(self.ctes[name.lower()]))
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
                if self._accept_word("UNKNOWN"):
                    # IS [NOT] UNKNOWN on a boolean = IS [NOT] NULL
                    e = ("call", "IS NOT NULL" if neg else "IS NULL", [e])
                    continue
                tv = self.accept_kw("TRUE", "FALSE")
                if tv is not None:
                    # x IS [NOT] TRUE/FALSE: NULL counts as "not true" and
                    # "not false" (three-valued IS, reference
                    # rex/core/call.py IsTrue/IsFalse lowering)
                    want = ("call", "AND", [("call", "IS NOT NULL", [e]),
                                            e if tv == "TRUE"
                                            else ("call", "NOT", [e])])
                    e = ("call", "NOT", [want]) if neg else want
                    continue
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
                e = self._like_tail(e, "LIKE")
                continue
            if self._accept_word("ILIKE"):
                e = self._like_tail(e, "ILIKE")
                continue
            if self._accept_word("SIMILAR"):
                self._expect_word("TO")
                e = self._like_tail(e, "SIMILAR")
                continue
            if self.accept_kw("NOT"):
                if self.accept_kw("IN"):
                    e = ("call", "NOT", [self._in_list(e)])
                    continue
                if self.accept_kw("LIKE"):
                    e = ("call", "NOT", [self._like_tail(e, "LIKE")])
                    continue
                if self._accept_word("ILIKE"):
                    e = ("call", "NOT", [self._like_tail(e, "ILIKE")])
                    continue
                if self._accept_word("SIMILAR"):
                    self._expect_word("TO")
                    e = ("call", "NOT", [self._like_tail(e, "SIMILAR")])
                    continue
                if self.accept_kw("BETWEEN"):
                    lo = self.add_expr()
                    self.expect_kw("AND")
                    hi = self.add_expr()
                    e = ("call", "NOT",
                         [("call", "AND",
                           [("call", ">=", [e, lo]), ("call", "<=", [e, hi])])])
                    continue
                raise ValueError(
                    "expected IN/BETWEEN/LIKE/ILIKE/SIMILAR after NOT")
            if self.accept_kw("IN"):
                e = self._in_list(e)
                continue
            break
        return e

    def _like_tail(self, e, op):
        """LIKE/ILIKE/SIMILAR TO pattern [ESCAPE '<c>'] — the escape char
        rides as a third literal operand (reference rex/core/call.py LIKE
        lowering takes an escape argument)."""
        pat = self.add_expr()
        args = [e, pat]
        if self._accept_word("ESCAPE"):
            esc = self.next()
            if esc[0] != "str":
                raise ValueError("ESCAPE needs a string literal")
            args.append(("lit", esc[1], "VARCHAR"))
        return ("call", op, args)

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
            if t[1] in ("FIRST", "LAST") \
                    and self.toks[self.i + 1] == ("op", "("):
                # FIRST(x) aggregate — the word is reserved for NULLS
                # FIRST/LAST but callable as a function name
                self.next()
                return self._func_call(t[1])
            if t[1] in ("DATE", "TIMESTAMP"):
                # DATE '2026-01-01' is a literal; a bare DATE/TIMESTAMP not
                # followed by a string is a column reference (a column
                # aliased `AS date` being read back)
                self.next()
                if self.peek()[0] == "str":
                    return ("lit", self.next()[1], t[1])
                return self._colref_tail(t[1].lower())
            if t[1] == "INTERVAL":
                # INTERVAL '<n>' DAY|WEEK|MONTH|YEAR, or the single-string
                # form INTERVAL '5 days' (reference test_rex.py interval
                # arithmetic)
                self.next()
                s = self.next()
                if s[0] not in ("str", "num"):
                    raise ValueError("INTERVAL needs a quantity literal")
                txt = str(s[1]).strip()
                parts = txt.split()
                units = ("DAY", "WEEK", "MONTH", "YEAR", "QUARTER",
                         "HOUR", "MINUTE", "SECOND", "MILLISECOND",
                         "MICROSECOND")
                if s[0] == "str" and len(parts) == 2                         and parts[1].upper().rstrip("S") in units:
                    return ("interval", int(parts[0]),
                            parts[1].upper().rstrip("S"))
                u = self.next()
                unit = str(u[1]).upper().rstrip("S")
                if unit not in units:
                    raise ValueError(f"INTERVAL unit {u[1]!r} not supported")
                return ("interval", int(txt), unit)
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
            if name.upper() in ("CURRENT_TIMESTAMP", "CURRENT_DATE",
                                "CURRENT_TIME", "LOCALTIME",
                                "LOCALTIMESTAMP", "NOW"):
                # parenthesis-less niladic datetime functions (reference
                # call.py maps all of them to pd.Timestamp.now())
                if self.peek() == ("op", "("):
                    self.next()
                    self.expect_op(")")
                return ("call", "CURRENT_TIMESTAMP", [])
            if name.upper() == "DECIMAL" and self.peek()[0] == "str":
                # DECIMAL '100.2' typed literal (reference maps DECIMAL to
                # float64, mappings.py SqlTypeName.DECIMAL)
                return ("lit", float(self.next()[1]), "DOUBLE")
            if name.upper() == "TIME" and self.peek()[0] == "str":
                # TIME 'HH:MM:SS[.fff]' → epoch-day timestamp (reference
                # test_rex.py maps TIME to 1970-01-01 + time-of-day)
                txt = self.next()[1].strip()
                hh, mm, ss = txt.split(":")
                ns = (int(hh) * 3600 + int(mm) * 60) * 1_000_000_000
                if "." in ss:
                    sec, frac = ss.split(".")
                    ns += int(sec) * 1_000_000_000
                    ns += int((frac + "0" * 9)[:9])
                else:
                    ns += int(ss) * 1_000_000_000
                return ("lit", ns, "TIMESTAMP")
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
                if field == "CENTURIES":
                    field = "CENTURY"
                elif field.endswith("S") and field != "DOW":
                    field = field[:-1]
                if field in ("MILLENIUM", "MILLENNIUM"):
                    field = "MILLENNIUM"  # reference accepts the misspelling
                if field not in ("YEAR", "MONTH", "DAY", "HOUR", "MINUTE",
                                 "SECOND", "DATE", "CENTURY", "DECADE",
                                 "MILLENNIUM", "DOW", "DOY", "QUARTER",
                                 "MICROSECOND", "MILLISECOND", "WEEK"):
                    raise ValueError(f"EXTRACT({field}) not supported")
                self.expect_kw("FROM")
                e = self.expr()
                self.expect_op(")")
                return ("call", f"EXTRACT_{field}", [e])
            if self.peek() == ("op", "("):
                return self._func_call(name)
            return self._colref_tail(name)
        raise ValueError(f"unexpected token {t}")

    def _colref_tail(self, name):
        if self.accept_op("."):
            if self.accept_op("*"):
                return ("qstar", name)  # t.* — qualified wildcard
            col = self._name()
            return ("col", name, col)
        return ("col", None, name)

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

    _TS_UNITS = ("YEAR", "QUARTER", "MONTH", "WEEK", "DAY", "HOUR",
                 "MINUTE", "SECOND", "MILLISECOND", "MICROSECOND")

    def _func_call(self, name):
        fname = name.upper()
        self.expect_op("(")
        if fname in ("TIMESTAMPADD", "TIMESTAMPDIFF"):
            # TIMESTAMPADD(unit, n, ts) — the unit rides as a bare word
            # (reference rex/core/call.py DatetimeSubOperation /
            # TimeStampAddOperation)
            ut = self.next()
            unit = str(ut[1]).upper()
            if unit not in self._TS_UNITS:
                raise ValueError(f"{fname} unit {ut[1]!r} not supported")
            self.expect_op(",")
            a1 = self.expr()
            self.expect_op(",")
            a2 = self.expr()
            self.expect_op(")")
            if fname == "TIMESTAMPDIFF":
                return ("call", "TIMESTAMPDIFF",
                        [("lit", unit, "VARCHAR"), a1, a2])
            # fold a literal count into an interval so the existing
            # date±INTERVAL machinery does the arithmetic
            def _fold_int(a):
                if not isinstance(a, tuple):
                    return None
                if a[0] == "lit" and isinstance(a[1], int):
                    return a[1]
                if a[0] == "call" and a[1] == "NEG":
                    v = _fold_int(a[2][0])
                    return None if v is None else -v
                if a[0] == "call" and a[1] in ("+", "-", "*") \
                        and len(a[2]) == 2:
                    x_, y_ = _fold_int(a[2][0]), _fold_int(a[2][1])
                    if x_ is None or y_ is None:
                        return None
                    return {"+": x_ + y_, "-": x_ - y_,
                            "*": x_ * y_}[a[1]]
                return None

            n = _fold_int(a1)
            if n is None:
                raise ValueError(
                    f"{fname} count must be an integer literal")
            return ("call", "+", [a2, ("interval", n, unit)])
        if fname in ("LTRIM", "RTRIM", "BTRIM"):
            # mode-fixed trims (reference call.py TrimOperation variants)
            e = self.expr()
            ch = " "
            if self.accept_op(","):
                t = self.next()
                if t[0] != "str":
                    raise ValueError(f"{fname} trim set must be a string")
                ch = t[1]
            self.expect_op(")")
            mode = {"LTRIM": "LEADING", "RTRIM": "TRAILING",
                    "BTRIM": "BOTH"}[fname]
            return ("call", "TRIM", [e, ("lit", mode, "VARCHAR"),
                                     ("lit", ch, "VARCHAR")])
        if fname == "POSITION":
            # POSITION(needle IN hay [FROM start]) — Calcite form
            # needle parses below the IN-postfix level so `'a' IN a` is
            # read as the POSITION separator, not a membership test
            needle = self.add_expr()
            if self.accept_kw("IN"):
                hay = self.add_expr()
                args = [hay, needle]
                if self._accept_word("FROM"):
                    args.append(self.expr())
                self.expect_op(")")
                return ("call", "POSITION", args)
            args = [needle]
            while self.accept_op(","):
                args.append(self.expr())
            self.expect_op(")")
            return ("call", "POSITION", args)
        if fname == "TO_TIMESTAMP":
            # second argument is a strptime format; the reference writes it
            # double-quoted, which lexes as an identifier — accept both
            args = [self.expr()]
            if self.accept_op(","):
                t = self.next()
                if t[0] not in ("str", "id"):
                    raise ValueError("to_timestamp format must be a string")
                args.append(("lit", t[1], "VARCHAR"))
            self.expect_op(")")
            return ("call", "TO_TIMESTAMP", args)
        if fname == "OVERLAY":
            # OVERLAY(x PLACING y FROM n [FOR m]) — Calcite form
            x_ = self.expr()
            self._expect_word("PLACING")
            y_ = self.add_expr()
            self._expect_word("FROM")
            args = [x_, y_, self.add_expr()]
            if self._accept_word("FOR"):
                args.append(self.add_expr())
            self.expect_op(")")
            return ("call", "OVERLAY", args)
        if fname in ("FLOOR", "CEIL", "CEILING"):
            # FLOOR(x TO DAY) — datetime truncation form
            e = self.expr()
            if self._accept_word("TO"):
                ut = self.next()
                unit = str(ut[1]).upper()
                if unit not in self._TS_UNITS:
                    raise ValueError(f"{fname}(.. TO {ut[1]!r}) unsupported")
                self.expect_op(")")
                base = "CEIL" if fname in ("CEIL", "CEILING") else "FLOOR"
                return ("call", f"{base}_TO_{unit}", [e])
            args = [e]
            while self.accept_op(","):
                args.append(self.expr())
            self.expect_op(")")
            return ("call", "CEIL" if fname == "CEILING" else fname, args)
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
                    nf = None  # engine default (NULLS LAST placement)
                    if self.accept_kw("NULLS"):
                        nf = self.accept_kw("FIRST") is not None
                        if not nf:
                            self.expect_kw("LAST")
                    order.append((e, desc, nf))
                    if not self.accept_op(","):
                        break
            frame = None
            fk = "ROWS" if self._accept_word("ROWS") else (
                "RANGE" if self._accept_word("RANGE") else None)
            if fk is not None:
                def bound():
                    if self._accept_word("UNBOUNDED"):
                        side = "preceding" \
                            if self._accept_word("PRECEDING") else \
                            ("following" if self._accept_word("FOLLOWING")
                             else None)
                        if side is None:
                            raise ValueError("UNBOUNDED needs PRECEDING/"
                                             "FOLLOWING")
                        return ("unbounded_" + side, None)
                    if self._accept_word("CURRENT"):
                        self._expect_word("ROW")
                        return ("current", 0)
                    t = self.next()
                    if t[0] != "num":
                        raise ValueError(f"bad frame bound {t}")
                    k = int(t[1])
                    if self._accept_word("PRECEDING"):
                        return ("preceding", k)
                    self._expect_word("FOLLOWING")
                    return ("following", k)

                if self.accept_kw("BETWEEN"):
                    lo = bound()
                    self.expect_kw("AND")
                    hi = bound()
                else:
                    lo = bound()
                    hi = ("current", 0)
                frame = (fk.lower(), lo, hi)
                # the RANGE spelling of the default frame IS the default
                if frame == ("range", ("unbounded_preceding", None),
                             ("current", 0)):
                    frame = None
            self.expect_op(")")
            return ("window", fname.lower(), args, tuple(part),
                    tuple(order), frame)
        if fname in ("REGR_COUNT", "REGR_SXX", "REGR_SYY", "COVAR_POP",
                     "COVAR_SAMP") and len(args) == 2:
            # bivariate aggregates over non-NULL PAIRS, rewritten onto the
            # existing SUM/COUNT machinery (reference rel/custom/wrappers +
            # datafusion REGR lowering): REGR_COUNT = pairwise COUNT,
            # REGR_SXX/SYY = S_vv - S_v^2/N, COVAR = (S_yx - S_y S_x/N)/N
            y, x = args
            pair = ("call", "AND", [("call", "IS NOT NULL", [y]),
                                    ("call", "IS NOT NULL", [x])])
            filt = pair if filter_expr is None else                 ("call", "AND", [filter_expr, pair])
            cnt = ("agg", "count", [("star",)], False, filt)
            if fname == "REGR_COUNT":
                return cnt
            nf = ("cast", cnt, "DOUBLE")

            def sm(e):
                return ("agg", "sum", [e], False, filt)

            if fname in ("REGR_SXX", "REGR_SYY"):
                v = x if fname == "REGR_SXX" else y
                return ("call", "-",
                        [sm(("call", "*", [v, v])),
                         ("call", "/",
                          [("call", "*", [sm(v), sm(v)]), nf])])
            cov = ("call", "-",
                   [sm(("call", "*", [y, x])),
                    ("call", "/", [("call", "*", [sm(y), sm(x)]), nf])])
            den = nf if fname == "COVAR_POP" else                 ("call", "-", [nf, ("lit", 1, "BIGINT")])
            return ("call", "/", [cov, den])
        _AGG_ALIASES = {"MEAN": "avg", "STD": "stddev",
                        "STDDEVPOP": "stddev_pop",
                        "STDDEVSAMP": "stddev_samp",
                        "VARIANCE_POP": "var_pop",
                        "VARIANCEPOP": "var_pop",
                        "FIRST": "single_value"}
        if fname in _AGG_ALIASES:  # reference AGGREGATION_MAPPING aliases
            return ("agg", _AGG_ALIASES[fname], args, distinct,
                    filter_expr)
        if fname in AGG_FUNCS:
            return ("agg", fname.lower(), args, distinct, filter_expr)
        return ("call", fname, args)


def parse_sql(sql: str) -> SelectStmt:
    return Parser(sql.strip().rstrip(";")).parse()
