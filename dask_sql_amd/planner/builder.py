"""AST → LogicalPlan builder (name resolution, join shaping, aggregate
extraction). Produces the plan shapes the physical plugins consume
(SURVEY.md §8b); semantics pinned end-to-end by the golden tests.

Mirrors, functionally, what the reference gets from DataFusion SqlToRel +
the optimizer rule subset that matters on this path (PushDownFilter /
EliminateCrossJoin — src/sql/optimizer.rs:53-98): single-table conjuncts are
pushed below joins, comma-joins with WHERE equalities become inner joins.
"""
from __future__ import annotations

import numpy as np

from dask_sql_amd.planner.parser import SelectStmt, TableRef, parse_sql
from dask_sql_amd.planner.plan import (
    AggCall, AggregateNode, Call, Expression, Field, FilterNode, InputRef,
    JoinNode, Literal, LimitNode, LogicalPlan, ProjectionNode, RelDataType,
    SortNode, SqlType, TableScanNode,
)

_EPOCH = np.datetime64("1970-01-01")


def _date_to_days(s: str) -> int:
    return int((np.datetime64(s) - _EPOCH).astype(int))


_NUMERIC_RANK = {"TINYINT": 0, "SMALLINT": 1, "INTEGER": 2, "BIGINT": 3,
                 "DATE": 3, "FLOAT": 4, "DOUBLE": 5}


def _is_float(t: str) -> bool:
    return t in ("FLOAT", "DOUBLE")


def _common_type(a: str, b: str) -> str:
    if a == b:
        return a
    if a == "VARCHAR" or b == "VARCHAR":
        return "VARCHAR"
    ra, rb = _NUMERIC_RANK.get(a, 5), _NUMERIC_RANK.get(b, 5)
    return a if ra >= rb else b


def _add_months_host(is_date):
    """pandas DateOffset month/year arithmetic over raw day/ns ints (the
    UDF slow path hands operands over as host Series)."""
    def f(vals, months):
        import pandas as pd
        m = int(months.iloc[0])
        if is_date:
            base = pd.to_datetime(vals, unit="D")
            out = pd.Series(base + pd.DateOffset(months=m))
            res = ((out - pd.Timestamp(0)).dt.days).astype("float64")
            return res  # NaT → NaN → validity
        base = pd.to_datetime(vals)
        out = pd.Series(base + pd.DateOffset(months=m))
        if out.isna().any():
            res = out.to_numpy("datetime64[ns]").view("int64")                 .astype("float64")
            res[out.isna().to_numpy()] = np.nan
            return pd.Series(res)
        return pd.Series(out.to_numpy("datetime64[ns]").view("int64"))
    return f


def _isoweek_host(is_date):
    """EXTRACT(WEEK) = isocalendar().week (reference date_part WEEK)."""
    def f(vals):
        import pandas as pd
        base = pd.to_datetime(vals, unit="D") if is_date \
            else pd.to_datetime(vals)
        wk = pd.Series(base).dt.isocalendar().week.astype("float64")
        wk[pd.Series(base).isna().to_numpy()] = np.nan
        return wk
    return f


def _rand_host(integer):
    """RAND[_INTEGER] — reference call.py RandOperation (numpy RNG; a
    literal seed makes it deterministic)."""
    def f(*args):
        import numpy as np_
        import pandas as pd_
        if integer:
            high = args[0]
            seed = args[1] if len(args) > 1 else None
        else:
            high = None
            seed = args[0] if args else None
        n = len(args[0])
        sv = None
        if seed is not None and len(seed) and not pd_.isna(seed.iloc[0]):
            sv = int(seed.iloc[0])
        rng = np_.random.RandomState(sv)
        if integer:
            return pd_.Series(rng.randint(0, int(high.iloc[0]), n))
        return pd_.Series(rng.rand(n))
    return f


class Catalog:
    """What the builder needs from the Context's schema: table → fields,
    plus the registered scalar/aggregate UDFs (reference
    context.py:324/:415 register_function/register_aggregation — one
    shared namespace for both kinds)."""

    def __init__(self):
        self.tables: dict[str, list[tuple[str, str]]] = {}
        self.views: dict[str, str] = {}  # name → view SQL (re-planned
        # at every use, like the reference's CreateView persist=False)
        self.functions: dict[str, tuple] = {}     # name → (f, ret_sql, row_udf)
        self.aggregations: dict[str, tuple] = {}  # name → (obj, ret_sql)
        # hidden builtins: calendar month arithmetic on COLUMNS executes
        # host-side exactly as the reference does (pandas Timestamp +
        # DateOffset, rex/core/call.py datetime ops) through the normal
        # UDF machinery
        self.functions["__add_months_ts__"] = (
            _add_months_host(False), "TIMESTAMP", False,
            [("x", "TIMESTAMP"), ("m", "BIGINT")])
        self.functions["__add_months_date__"] = (
            _add_months_host(True), "DATE", False,
            [("x", "DATE"), ("m", "BIGINT")])
        self.functions["__isoweek_ts__"] = (
            _isoweek_host(False), "BIGINT", False, [("x", "TIMESTAMP")])
        self.functions["__isoweek_date__"] = (
            _isoweek_host(True), "BIGINT", False, [("x", "DATE")])
        self.functions["__rand__"] = (
            _rand_host(False), "DOUBLE", False, [("seed", "BIGINT")])
        self.functions["__rand_integer__"] = (
            _rand_host(True), "BIGINT", False,
            [("high", "BIGINT"), ("seed", "BIGINT")])

    def add(self, name, fields):
        self.tables[name.lower()] = fields

    def drop(self, name):
        self.tables.pop(name.lower(), None)

    def get(self, name):
        key = name.lower()
        if key not in self.tables:
            raise KeyError(f"Table {name!r} not registered")
        return self.tables[key]


def _expr_type(e: Expression) -> str:
    t = e.getType()
    return t.getSqlType() if t else "DOUBLE"


class Builder:
    def __init__(self, catalog: Catalog, schema_name: str = "root"):
        self.catalog = catalog
        self.schema_name = schema_name

    # ------------------------------------------------------------------ API
    def build(self, sql: str) -> LogicalPlan:
        from dask_sql_amd.planner.prune import prune_plan
        stmt = parse_sql(sql)
        return prune_plan(self.build_stmt(stmt))

    # ----------------------------------------------------------------- scans
    def _scan(self, tr: TableRef) -> LogicalPlan:
        if getattr(tr, "subquery", None) is None and tr.name \
                and tr.name.lower() in self.catalog.views:
            # view reference: inline the stored SELECT as a derived table
            from dask_sql_amd.planner.parser import parse_sql as _ps
            sub = _ps(self.catalog.views[tr.name.lower()])
            tr = TableRef(name=None, alias=tr.alias or tr.name,
                          subquery=sub)
        if getattr(tr, "subquery", None) is not None:
            # derived table: build the sub-select, requalify its output with
            # the alias (reference: DataFusion subquery alias rel)
            sub = self.build_stmt(tr.subquery)
            qual = tr.alias.lower()
            fields = [Field(f.getName(), f.getType(), qualifier=qual)
                      for f in sub.getRowType().getFieldList()]
            named = [(InputRef(i, f.getType()), f.getName())
                     for i, f in enumerate(fields)]
            return LogicalPlan("Projection", [sub], RelDataType(fields),
                               ProjectionNode(named))
        fields_spec = self.catalog.get(tr.name)
        qual = (tr.alias or tr.name).lower()
        fields = [Field(n, SqlType(t), qualifier=qual) for n, t in fields_spec]
        node = TableScanNode(self.schema_name, tr.name.lower())
        return LogicalPlan("TableScan", [], RelDataType(fields), node)

    # ------------------------------------------------------- expr resolution
    def _resolve(self, ast, plan: LogicalPlan) -> Expression:
        fields = plan.getRowType().getFieldList()
        kind = ast[0]
        if kind == "col":
            _, q, n = ast
            matches = [
                i for i, f in enumerate(fields)
                if f.getName().lower() == n.lower()
                and (q is None or (f.qualifier or "").lower() == q.lower())
            ]
            if not matches:
                # also match already-qualified output names like "lhs.a"
                full = f"{q}.{n}".lower() if q else n.lower()
                matches = [i for i, f in enumerate(fields)
                           if f.getName().lower() == full]
            if not matches:
                raise KeyError(f"column {q + '.' if q else ''}{n} not found in "
                               f"{[f.getQualifiedName() for f in fields]}")
            if len(matches) > 1:
                # names differing only in case (quoted identifiers): an
                # EXACT-case match wins before declaring ambiguity
                exact = [i for i in matches if fields[i].getName() == n]
                if len(exact) == 1:
                    matches = exact
            if len(matches) > 1 and q is None:
                raise KeyError(f"ambiguous column {n}")
            i = matches[0]
            return InputRef(i, fields[i].getType())
        if kind == "lit":
            _, v, t = ast
            if t == "DATE":
                return Literal(_date_to_days(v), SqlType("DATE"))
            if t == "TIMESTAMP":
                ns = int(np.datetime64(v, "ns").astype("int64"))
                return Literal(ns, SqlType("TIMESTAMP"))
            if t == "NULL":
                return Literal(None, SqlType("NULL"))
            return Literal(v, SqlType(t))
        if kind == "interval":
            return Literal((ast[1], ast[2]), SqlType("INTERVAL"))
        if kind == "cast":
            _, sub, ty = ast
            e = self._resolve(sub, plan)
            return Call("CAST", [e], SqlType(ty))
        if kind == "case":
            _, whens, els = ast
            ops = []
            for cond, val in whens:
                ops.append(self._resolve(cond, plan))
                ops.append(self._resolve(val, plan))
            ops.append(self._resolve(els, plan) if els is not None
                       else Literal(None, SqlType("NULL")))
            ty = _expr_type(ops[1])
            return Call("CASE", ops, SqlType(ty))
        if kind == "call":
            _, op, args = ast
            ops = [self._resolve(a, plan) for a in args]
            fn = self.catalog.functions.get(op.lower())
            if fn is not None:
                return Call(f"UDF:{op.lower()}", ops, SqlType(fn[1]))
            if op == "CURRENT_TIMESTAMP" and not ops:
                # evaluated when the plan is BUILT (the reference evaluates
                # pd.Timestamp.now() when the rex runs — once per query
                # either way); _get_ral skips the plan cache for these
                import pandas as _pd
                return Literal(int(_pd.Timestamp.now().value),
                               SqlType("TIMESTAMP"))
            if op in ("RAND", "RANDOM", "RAND_INTEGER"):
                # reference call.py RandOperation/RandIntegerOperation —
                # host builtins (Python randomness IS the reference's
                # execution model); optional leading seed argument
                if op == "RAND_INTEGER":
                    if len(ops) == 2:
                        seed, high = ops
                    else:
                        seed, high = None, ops[0]
                    args = [high] + ([seed] if seed is not None else [])
                    return Call("UDF:__rand_integer__", args,
                                SqlType("BIGINT"))
                args = list(ops) if ops else [Literal(None,
                                                       SqlType("NULL"))]
                return Call("UDF:__rand__", args, SqlType("DOUBLE"))
            if op in ("DATEPART", "DATE_PART") and len(ops) == 2 \
                    and isinstance(ops[0], Literal):
                # DATEPART('field', x) — function form of EXTRACT
                # (reference call.py datepart → ExtractOperation)
                field = str(ops[0].getValue()).upper()
                if field == "CENTURIES":
                    field = "CENTURY"
                elif field.endswith("S"):
                    field = field[:-1]
                if field == "MILLENIUM":
                    field = "MILLENNIUM"
                # fall through: EXTRACT_WEEK and the generic typing below
                # treat this exactly like EXTRACT(field FROM x)
                ops = [ops[1]]
                op = f"EXTRACT_{field}"
            if op == "EXTRACT_WEEK":
                x = ops[0]
                name = "__isoweek_ts__" if _expr_type(x) == "TIMESTAMP" \
                    else "__isoweek_date__"
                return Call(f"UDF:{name}", [x], SqlType("BIGINT"))
            if op == "TO_TIMESTAMP":
                # reference rex/core/call.py ToTimestampOperation: string →
                # strptime (host fold); numeric → seconds since epoch; an
                # existing DATE/TIMESTAMP passes through (promoted to ns)
                x = ops[0]
                fmt = "%Y-%m-%d %H:%M:%S"
                if len(ops) > 1 and isinstance(ops[1], Literal):
                    fmt = str(ops[1].getValue()).replace('"', "")
                    fmt = fmt.replace("'", "")
                if isinstance(x, Literal) \
                        and isinstance(x.getValue(), str):
                    from datetime import datetime as _dt
                    ns = int(np.datetime64(_dt.strptime(x.getValue(), fmt),
                                           "ns").astype("int64"))
                    return Literal(ns, SqlType("TIMESTAMP"))
                tx = _expr_type(x)
                if tx == "TIMESTAMP":
                    return x
                if tx == "VARCHAR":
                    # string column: strptime runs once per dictionary
                    # entry (physical/rex.py dict_int_fn TO_TIMESTAMP)
                    return Call("TO_TIMESTAMP",
                                [x, Literal(fmt, SqlType("VARCHAR"))],
                                SqlType("TIMESTAMP"))
                if len(ops) > 1 and fmt != "%Y-%m-%d %H:%M:%S":
                    raise NotImplementedError(
                        "Integer input does not accept a format argument")
                scale = 86_400_000_000_000 if tx == "DATE" \
                    else 1_000_000_000
                mul = Call("*", [x, Literal(scale, SqlType("BIGINT"))],
                           SqlType("TIMESTAMP"))
                if tx in ("DOUBLE", "FLOAT", "DECIMAL"):
                    # fractional seconds survive the ns multiply; the CAST
                    # truncates to integer ns for the TIMESTAMP column
                    return Call("CAST", [mul], SqlType("TIMESTAMP"))
                return mul
            if op in ("DEGREES", "RADIANS", "LOG10", "CBRT", "SIGN",
                      "TRUNCATE") and len(ops) == 1:
                # rewrites onto existing scalar ops (reference
                # rex/core/call.py da.degrees/radians/log10/cbrt/sign/trunc)
                import math as _m
                x = ops[0]
                D = SqlType("DOUBLE")
                if op == "DEGREES":
                    return Call("*", [x, Literal(180.0 / _m.pi, D)], D)
                if op == "RADIANS":
                    return Call("*", [x, Literal(_m.pi / 180.0, D)], D)
                if op == "LOG10":
                    return Call("*", [Call("LN", [x], D),
                                      Literal(1.0 / _m.log(10.0), D)], D)
                zero = Literal(0, SqlType("BIGINT"))
                gt = Call(">", [x, zero], SqlType("BOOLEAN"))
                lt = Call("<", [x, zero], SqlType("BOOLEAN"))
                sgn_i = Call("-", [gt, lt], SqlType("BIGINT"))
                is_f = _expr_type(x) in ("DOUBLE", "FLOAT", "DECIMAL")
                if op == "SIGN":
                    if not is_f:
                        return sgn_i
                    # np.sign(NaN) = NaN
                    nan_guard = Call("<>", [x, x], SqlType("BOOLEAN"))
                    return Call("CASE", [nan_guard, x,
                                         Call("CAST", [sgn_i], D)], D)
                if op == "CBRT":
                    root = Call("POWER",
                                [Call("ABS", [x], D),
                                 Literal(1.0 / 3.0, D)], D)
                    return Call("*", [sgn_i, root], D)
                # TRUNCATE: toward zero (da.trunc)
                ge = Call(">=", [x, zero], SqlType("BOOLEAN"))
                return Call("CASE", [ge, Call("FLOOR", [x], D),
                                     Call("CEIL", [x], D)], D)
            if op in ("+", "-") and any(
                    isinstance(o, Literal)
                    and o.getType().getSqlType() == "INTERVAL" for o in ops):
                return self._date_interval(op, ops)
            if op in ("=", "<>", "<", "<=", ">", ">=") and len(ops) == 2:
                ta, tb = _expr_type(ops[0]), _expr_type(ops[1])
                # DATE/TIMESTAMP vs string literal: coerce the literal to
                # the temporal side (reference lets pandas compare
                # datetime64 against a parseable string)
                for i_, (tx_, ty_) in ((0, (ta, tb)), (1, (tb, ta))):
                    o = ops[i_]
                    if ty_ in ("DATE", "TIMESTAMP") and tx_ == "VARCHAR" \
                            and isinstance(o, Literal) \
                            and isinstance(o.getValue(), str):
                        try:
                            if ty_ == "DATE":
                                v = _date_to_days(o.getValue())
                            else:
                                v = int(np.datetime64(o.getValue(), "ns")
                                        .astype("int64"))
                        except Exception:
                            continue
                        ops[i_] = Literal(v, SqlType(ty_))
                ta, tb = _expr_type(ops[0]), _expr_type(ops[1])
                if {ta, tb} == {"TIMESTAMP", "DATE"}:
                    # promote the DATE side to ns so the compare is exact
                    def _prom(o):
                        if _expr_type(o) != "DATE":
                            return o
                        day_ns = 86_400_000_000_000
                        if isinstance(o, Literal):
                            return Literal(int(o.getValue()) * day_ns,
                                           SqlType("TIMESTAMP"))
                        return Call("*", [o, Literal(day_ns,
                                                     SqlType("BIGINT"))],
                                    SqlType("TIMESTAMP"))
                    ops = [_prom(o) for o in ops]
            if op in ("=", "<>", "<", "<=", ">", ">=", "AND", "OR", "NOT",
                      "IS NULL", "IS NOT NULL", "LIKE", "ILIKE", "SIMILAR"):
                ty = "BOOLEAN"
            elif op == "NEG":
                ty = _expr_type(ops[0])
            elif op in ("UPPER", "LOWER", "SUBSTRING", "SUBSTR", "CONCAT",
                        "TRIM", "REPLACE", "INITCAP", "OVERLAY"):
                ty = "VARCHAR"
            elif op in ("CHAR_LENGTH", "CHARACTER_LENGTH", "LENGTH",
                        "POSITION"):
                ty = "BIGINT"
            elif op in ("FLOOR", "CEIL", "CEILING", "ROUND", "EXP", "LN",
                        "LOG", "POWER", "POW", "SQRT", "SIN", "COS", "TAN",
                        "ASIN", "ACOS", "ATAN", "ATAN2", "COT"):
                ty = "DOUBLE"
            elif op == "EXTRACT_DATE":
                ty = "DATE"
            elif op == "MOD":
                ta = _expr_type(ops[0])
                tb = _expr_type(ops[1])
                ty = "DOUBLE" if "DOUBLE" in (ta, tb) or "FLOAT" in (
                    ta, tb) or "DECIMAL" in (ta, tb) else "BIGINT"
            elif op.startswith("EXTRACT_") or op in (
                    "YEAR", "MONTH", "DAY", "DAYOFMONTH"):
                ty = "BIGINT"
            elif op.startswith("FLOOR_TO_") or op.startswith("CEIL_TO_") \
                    or op == "LAST_DAY":
                ty = _expr_type(ops[0])
            elif op == "TIMESTAMPDIFF":
                # TIMESTAMPDIFF(unit, a, b) = truncated count of whole
                # units in b - a; sub-month units only (MONTH/YEAR need
                # calendar spans)
                unit = ops[0].getValue()
                ns = {"MICROSECOND": 1_000, "MILLISECOND": 1_000_000,
                      "SECOND": 1_000_000_000, "MINUTE": 60_000_000_000,
                      "HOUR": 3_600_000_000_000,
                      "DAY": 86_400_000_000_000,
                      "WEEK": 7 * 86_400_000_000_000}.get(unit)
                if ns is None:
                    raise NotImplementedError(
                        f"TIMESTAMPDIFF({unit}) needs calendar spans")
                def _ns(o):
                    if _expr_type(o) == "DATE":
                        return Call("*", [o, Literal(86_400_000_000_000,
                                                     SqlType("BIGINT"))],
                                    SqlType("TIMESTAMP"))
                    return o
                diff = Call("-", [_ns(ops[2]), _ns(ops[1])],
                            SqlType("BIGINT"))
                return Call("/", [diff, Literal(ns, SqlType("BIGINT"))],
                            SqlType("BIGINT"))
            elif op == "ABS":
                ty = _expr_type(ops[0])
            elif op == "/":
                ty = _common_type(_expr_type(ops[0]), _expr_type(ops[1]))
            else:
                ty = _common_type(_expr_type(ops[0]),
                                  _expr_type(ops[1]) if len(ops) > 1
                                  else _expr_type(ops[0]))
            return Call(op, ops, SqlType(ty))
        if kind == "scalar_sub":
            from dask_sql_amd.planner.plan import ScalarSub
            sub = self.build_stmt(ast[1])
            if len(sub.getRowType().getFieldList()) != 1:
                raise ValueError("scalar subquery must select one column")
            return ScalarSub(sub)
        if kind == "agg":
            raise ValueError("aggregate in non-aggregate position")
        raise ValueError(f"cannot resolve {ast!r}")

    def _semi_anti_join(self, plan, subplan, outer_asts, negated,
                        null_aware=False):
        """SEMI/ANTI join `plan` against the DISTINCT subplan output on
        positional key equalities (outer_asts resolve against the combined
        row; sub keys are the subplan's columns in order)."""
        sfields = subplan.getRowType().getFieldList()
        node = AggregateNode(
            [InputRef(i, f.getType()) for i, f in enumerate(sfields)],
            [], distinct_node=True,
            distinct_columns=[f.getName() for f in sfields])
        subplan = LogicalPlan("Distinct", [subplan], subplan.getRowType(),
                              node)
        lhs_fields = plan.getRowType().getFieldList()
        # outer key asts resolve in the OUTER scope only — a subplan output
        # sharing the column name (IN over a grouped projection of the same
        # table) must not make it ambiguous
        tmp = LogicalPlan("__combined__", [], RelDataType(lhs_fields),
                          None)
        cond = None
        for i, ast in enumerate(outer_asts):
            eq = Call("=", [self._resolve(ast, tmp),
                            InputRef(len(lhs_fields) + i,
                                     sfields[i].getType())],
                      SqlType("BOOLEAN"))
            cond = eq if cond is None else Call("AND", [cond, eq],
                                                SqlType("BOOLEAN"))
        jt = "LEFTANTI" if negated else "LEFTSEMI"
        node = JoinNode(jt, cond)
        # SQL NOT IN: a NULL anywhere in the subquery output makes the
        # predicate non-TRUE for EVERY row (three-valued logic) — the
        # plugin returns an empty result when the build side holds NULL
        # keys. Closes the r1 documented divergence.
        node.null_aware = null_aware
        return LogicalPlan("Join", [plan, subplan], RelDataType(lhs_fields),
                           node)

    def _decorrelate_exists(self, sub, lead_items=None, tail_items=None,
                            group=False, key_prefix=""):
        """Split the EXISTS subquery's WHERE into local conjuncts and
        equality correlations on OUTER columns (qualified names not bound
        by the sub's own FROM). Returns (subplan selecting the inner keys,
        [outer key asts]), or (None, None) when uncorrelated. Only
        equality correlation with qualified outer references is supported
        (what DataFusion's decorrelation handles for the reference)."""
        sub_quals = {(t.alias or t.name).lower() for t in sub.from_tables
                     if (t.alias or t.name)}
        sub_quals |= {(j.table.alias or j.table.name).lower()
                      for j in sub.joins if (j.table.alias or j.table.name)}
        # column names BOUND by the sub's own FROM (registered tables; a
        # derived-table member makes the set unknowable → conservative:
        # unqualified names stay local, only qualified correlation works)
        local_cols = set()
        local_known = True
        for t in list(sub.from_tables) + [j.table for j in sub.joins]:
            if getattr(t, "subquery", None) is not None or t.name is None:
                local_known = False
                continue
            try:
                local_cols |= {n.lower() for n, _ in
                               self.catalog.get(t.name)}
            except KeyError:
                local_known = False
        conjs = self._conjuncts(sub.where)

        def outer_col(x):
            if not (isinstance(x, tuple) and x[0] == "col"):
                return False
            if x[1] is not None:
                return x[1].lower() not in sub_quals
            # unqualified outer reference (TPC-H Q2/Q20 style): a name the
            # sub's own tables do not bind
            return local_known and x[2].lower() not in local_cols

        def has_outer(ast):
            for q, n in self._tables_of(ast):
                if q is not None and q.lower() not in sub_quals:
                    return True
                if q is None and local_known \
                        and n.lower() not in local_cols:
                    return True
            return False

        local, inner_keys, outer_keys = [], [], []
        for cj in conjs:
            if not has_outer(cj):
                local.append(cj)
                continue
            if not (cj[0] == "call" and cj[1] == "=" and len(cj[2]) == 2):
                raise NotImplementedError(
                    "EXISTS correlation must be qualified equality "
                    "(outer.col = inner.col)")
            a, b = cj[2]
            if outer_col(a) and not has_outer(b):
                outer_keys.append(a)
                inner_keys.append(b)
            elif outer_col(b) and not has_outer(a):
                outer_keys.append(b)
                inner_keys.append(a)
            else:
                raise NotImplementedError(
                    "EXISTS correlation must pair one outer column with an "
                    "inner expression")
        if not outer_keys:
            return None, None
        where = None
        for cj in local:
            where = cj if where is None else ("call", "AND", [where, cj])
        items = list(lead_items or [])
        items += [(k, f"{key_prefix}ck{i}") for i, k in
                  enumerate(inner_keys)]
        items += list(tail_items or [])
        s2 = SelectStmt(items=items, from_tables=sub.from_tables,
                        joins=sub.joins, where=where,
                        group_by=list(inner_keys) if group else [])
        return self.build_stmt(s2), outer_keys

    @staticmethod
    def _date_interval(op, ops):
        """date ± INTERVAL: DAY/WEEK fold to day-int arithmetic; MONTH/YEAR
        use exact calendar math on literal dates (the reference gets this
        from pandas Timestamp + DateOffset, rex/core/call.py datetime ops)."""
        iv = next(o for o in ops
                  if isinstance(o, Literal)
                  and o.getType().getSqlType() == "INTERVAL")
        other = ops[0] if ops[1] is iv else ops[1]
        if ops[0] is iv:
            raise NotImplementedError("INTERVAL on the left of +/-")
        n_, unit = iv.getValue()
        sign = 1 if op == "+" else -1
        is_ts = _expr_type(other) == "TIMESTAMP"
        sub_day = {"HOUR": 3_600_000_000_000, "MINUTE": 60_000_000_000,
                   "SECOND": 1_000_000_000, "MILLISECOND": 1_000_000,
                   "MICROSECOND": 1_000}
        if unit in sub_day:
            if not is_ts:
                raise NotImplementedError(
                    f"{unit} interval on a DATE operand")
            step = n_ * sub_day[unit]
            if isinstance(other, Literal):
                return Literal(int(other.getValue()) + sign * step,
                               SqlType("TIMESTAMP"))
            return Call(op, [other, Literal(step, SqlType("BIGINT"))],
                        SqlType("TIMESTAMP"))
        if unit == "QUARTER":
            n_, unit = n_ * 3, "MONTH"
        if unit in ("DAY", "WEEK"):
            days = n_ * (7 if unit == "WEEK" else 1)
            step = days * (86_400_000_000_000 if is_ts else 1)
            out_t = "TIMESTAMP" if is_ts else "DATE"
            if isinstance(other, Literal):
                return Literal(int(other.getValue()) + sign * step,
                               SqlType(out_t))
            return Call(op, [other, Literal(step, SqlType("BIGINT"))],
                        SqlType(out_t))
        if not isinstance(other, Literal):
            # column operand: exact calendar arithmetic on the host UDF
            # path (reference pandas + DateOffset)
            months = sign * n_ * (12 if unit == "YEAR" else 1)
            name = "__add_months_ts__" if is_ts else "__add_months_date__"
            return Call(f"UDF:{name}",
                        [other, Literal(months, SqlType("BIGINT"))],
                        SqlType("TIMESTAMP" if is_ts else "DATE"))
        if is_ts:
            import pandas as pd
            months = sign * n_ * (12 if unit == "YEAR" else 1)
            nd = pd.Timestamp(int(other.getValue())) + \
                pd.DateOffset(months=months)
            return Literal(int(nd.value), SqlType("TIMESTAMP"))
        import calendar
        import datetime
        d = datetime.date(1970, 1, 1) + datetime.timedelta(
            days=int(other.getValue()))
        months = sign * n_ * (12 if unit == "YEAR" else 1)
        y = d.year + (d.month - 1 + months) // 12
        m = (d.month - 1 + months) % 12 + 1
        dd = min(d.day, calendar.monthrange(y, m)[1])
        nd = datetime.date(y, m, dd)
        return Literal((nd - datetime.date(1970, 1, 1)).days, SqlType("DATE"))

    # -------------------------------------------------------- ast utilities
    @staticmethod
    def _conjuncts(ast):
        if ast is None:
            return []
        if ast[0] == "call" and ast[1] == "AND":
            out = []
            for a in ast[2]:
                out.extend(Builder._conjuncts(a))
            return out
        return [ast]

    @staticmethod
    def _tables_of(ast, out=None):
        """qualifiers/col names referenced by an AST."""
        if out is None:
            out = []
        if not isinstance(ast, tuple):
            return out
        if ast[0] == "col":
            out.append((ast[1], ast[2]))
            return out
        if ast[0] == "call":
            for a in ast[2]:
                Builder._tables_of(a, out)
        elif ast[0] == "agg":
            for a in ast[2]:
                Builder._tables_of(a, out)
            if ast[4] is not None:
                Builder._tables_of(ast[4], out)
        elif ast[0] == "cast":
            Builder._tables_of(ast[1], out)
        elif ast[0] == "case":
            for c, v in ast[1]:
                Builder._tables_of(c, out)
                Builder._tables_of(v, out)
            if ast[2] is not None:
                Builder._tables_of(ast[2], out)
        return out

    def _refs_only(self, ast, quals: set, plan) -> bool:
        """every column in ast resolvable within plan (whose fields carry
        qualifiers in quals)"""
        for q, n in self._tables_of(ast):
            found = False
            for f in plan.getRowType().getFieldList():
                if f.getName().lower() == n.lower() and (
                    q is None or (f.qualifier or "").lower() == q.lower()
                ):
                    found = True
                    break
            if not found:
                return False
        return True

    @staticmethod
    def _has_agg(ast) -> bool:
        if not isinstance(ast, tuple):
            return False
        if ast[0] == "agg":
            return True
        if ast[0] == "call":
            return any(Builder._has_agg(a) for a in ast[2])
        if ast[0] == "cast":
            return Builder._has_agg(ast[1])
        if ast[0] == "case":
            return any(
                Builder._has_agg(c) or Builder._has_agg(v) for c, v in ast[1]
            ) or (ast[2] is not None and Builder._has_agg(ast[2]))
        return False

    # ------------------------------------------------------------- pipeline
    def _rewrite_udf_aggs(self, ast):
        """Registered aggregate UDFs parse as plain calls (the parser's
        AGG_FUNCS set is static) — rewrite them into agg nodes so they go
        through the Aggregate plan node like the reference's
        register_aggregation does."""
        if not isinstance(ast, tuple):
            return ast
        if ast[0] == "call" and isinstance(ast[1], str)                 and ast[1].lower() in self.catalog.aggregations:
            return ("agg", f"udf:{ast[1].lower()}",
                    [self._rewrite_udf_aggs(a) for a in ast[2]], False, None)
        if ast[0] == "call":
            return (ast[0], ast[1],
                    [self._rewrite_udf_aggs(a) for a in ast[2]])
        return ast

    def build_stmt(self, stmt) -> LogicalPlan:
        from dask_sql_amd.planner.parser import UnionStmt
        if isinstance(stmt, UnionStmt):
            return self._build_union(stmt)
        if self.catalog.aggregations:
            stmt.items = [(self._rewrite_udf_aggs(e), a)
                          for e, a in stmt.items]
        where_conjuncts = self._conjuncts(stmt.where)
        # IN (SELECT ...) conjuncts → SEMI/ANTI joins (DataFusion's subquery
        # decorrelation on the reference side). Pulled out before pushdown.
        in_subs = []
        exists_subs = []
        rest = []
        for cj in where_conjuncts:
            neg = (cj[0] == "call" and cj[1] == "NOT"
                   and isinstance(cj[2][0], tuple))
            base = cj[2][0] if neg else cj
            if base[0] == "in_sub":
                in_subs.append((base[1], base[2], neg))
            elif base[0] == "exists":
                exists_subs.append((base[1], neg))
            else:
                rest.append(cj)
        where_conjuncts = rest
        used = [False] * len(where_conjuncts)

        # Pushing a WHERE conjunct below a scan is only valid when that table
        # sits on the PRESERVED side of every join above it (DataFusion's
        # PushDownFilter rule pushes only to preserved sides; a predicate on
        # a null-supplying side must stay a post-join Filter, else
        # NULL-extended rows that SQL excludes would be returned).
        join_types = [jc.join_type for jc in stmt.joins]
        lhs_preserved = not any(t in ("RIGHT", "FULL") for t in join_types)
        # sql.predicate_pushdown=False disables the PushDownFilter analog
        # entirely (reference sql.yaml:predicate_pushdown; the WHERE then
        # applies as one post-join Filter — results identical)
        from dask_sql_amd import config as _config
        _pushdown = bool(_config.get("sql.predicate_pushdown", True))
        lhs_preserved = lhs_preserved and _pushdown

        def _has_ssub(ast):
            if not isinstance(ast, tuple):
                return False
            if ast[0] == "scalar_sub":
                return True
            if ast[0] == "call":
                return any(_has_ssub(a) for a in ast[2])
            if ast[0] == "cast":
                return _has_ssub(ast[1])
            if ast[0] == "case":
                return any(_has_ssub(x) for c, v in ast[1]
                           for x in (c, v)) or (
                    ast[2] is not None and _has_ssub(ast[2]))
            return False

        # 1. scans (+ pushed-down single-table filters, à la PushDownFilter)
        def scan_with_filters(tr: TableRef, push: bool = True) -> LogicalPlan:
            plan = self._scan(tr)
            if not push:
                return plan
            quals = {(tr.alias or tr.name).lower()}
            conds = []
            for i, cj in enumerate(where_conjuncts):
                if used[i] or self._has_agg(cj) or _has_ssub(cj):
                    # scalar subqueries resolve above the joins (correlated
                    # ones LEFT-join a grouped subplan) — never pushed
                    continue
                refs = self._tables_of(cj)
                if not refs:
                    continue  # scalar conditions stay global
                if self._refs_only(cj, quals, plan):
                    conds.append(cj)
                    used[i] = True
            for cj in conds:
                cond = self._resolve(cj, plan)
                plan = LogicalPlan("Filter", [plan], plan.getRowType(),
                                   FilterNode(cond))
            return plan

        tables = list(stmt.from_tables)
        if tables:
            plan = scan_with_filters(tables[0], push=lhs_preserved)
        else:
            # FROM-less SELECT (constants only, e.g. SELECT 1 + 1 —
            # reference supports via a one-row relation)
            plan = LogicalPlan("Values", [], RelDataType([]), None)

        def join_plans(lhs, rhs, join_type, cond_ast_list, on_expr_ast):
            lhs_fields = lhs.getRowType().getFieldList()
            rhs_fields = rhs.getRowType().getFieldList()
            combined = RelDataType(lhs_fields + rhs_fields)
            tmp = LogicalPlan("__combined__", [], combined, None)
            cond = None
            if on_expr_ast is not None:
                cond = self._resolve(on_expr_ast, tmp)
            elif cond_ast_list:
                ast = cond_ast_list[0]
                for c in cond_ast_list[1:]:
                    ast = ("call", "AND", [ast, c])
                cond = self._resolve(ast, tmp)
            if join_type in ("LEFTSEMI", "LEFTANTI"):
                out_fields = lhs_fields
            else:
                out_fields = lhs_fields + rhs_fields
            # duplicate base names across the two sides would collapse in the
            # name→backend mapping downstream: qualify them (what DataFusion's
            # qualified join fields give the reference; context.py:890-898)
            from collections import Counter
            counts = Counter(f.getName().lower() for f in out_fields)
            fixed = []
            for f in out_fields:
                if counts[f.getName().lower()] > 1 and f.qualifier:
                    fixed.append(Field(f"{f.qualifier}.{f.getName()}",
                                       f.getType(), qualifier=f.qualifier))
                else:
                    fixed.append(f)
            return LogicalPlan("Join", [lhs, rhs], RelDataType(fixed),
                               JoinNode(join_type, cond))

        # comma-joined tables: EliminateCrossJoin — find WHERE equalities
        for tr in tables[1:]:
            rhs = scan_with_filters(tr, push=lhs_preserved)
            lhs_fields = plan.getRowType().getFieldList()
            combined = RelDataType(lhs_fields + rhs.getRowType().getFieldList())
            tmp = LogicalPlan("__combined__", [], combined, None)
            conds = []
            if lhs_preserved:
                for i, cj in enumerate(where_conjuncts):
                    if used[i] or self._has_agg(cj) or _has_ssub(cj):
                        continue
                    refs = self._tables_of(cj)
                    if not refs:
                        continue
                    if self._refs_only(cj, set(), tmp) and not self._refs_only(
                        cj, set(), plan
                    ):
                        conds.append(cj)
                        used[i] = True
            plan = join_plans(plan, rhs, "INNER" if conds else "CROSS",
                              conds, None)

        # explicit JOIN clauses
        for k, jc in enumerate(stmt.joins):
            # rhs is preserved for INNER/RIGHT/CROSS; null-supplying under
            # LEFT/FULL (and semi/anti rhs columns are not in scope for
            # WHERE at all) — and a later RIGHT/FULL join null-supplies the
            # whole accumulated lhs, rhs included.
            rhs_push = _pushdown and jc.join_type not in (
                "LEFT", "FULL", "LEFTSEMI", "LEFTANTI"
            ) and not any(t in ("RIGHT", "FULL") for t in join_types[k + 1:])
            rhs = scan_with_filters(jc.table, push=rhs_push)
            plan = join_plans(plan, rhs, jc.join_type, [], jc.on)

        # correlated scalar subqueries with equality correlation (e.g.
        # `(SELECT MAX(x) FROM u WHERE u.k = t.k)`): rewrite to a GROUPED
        # subplan LEFT-JOINed on the correlation keys — DataFusion's
        # scalar-subquery decorrelation on the reference side (round-1
        # raised on these). The sub's single item must be an aggregate.
        n_user_fields = len(plan.getRowType().getFieldList())
        ssub_n = [0]

        def _pull_ssubs(ast):
            nonlocal plan
            if not isinstance(ast, tuple):
                return ast
            if ast[0] == "scalar_sub":
                sub = ast[1]

                def _has_agg(a):
                    if not isinstance(a, tuple):
                        return False
                    if a[0] == "agg":
                        return True
                    if a[0] == "call":
                        return any(_has_agg(x) for x in a[2])
                    if a[0] == "cast":
                        return _has_agg(a[1])
                    return False

                if (len(sub.items) == 1
                        and isinstance(sub.items[0][0], tuple)
                        and _has_agg(sub.items[0][0])
                        and not sub.group_by):
                    idx = ssub_n[0]
                    valname = f"__ssub{idx}"
                    try:
                        subplan, outer_keys = self._decorrelate_exists(
                            sub, tail_items=[(sub.items[0][0], valname)],
                            group=True, key_prefix=f"__ssub{idx}_")
                    except NotImplementedError:
                        return ast
                    if subplan is None:
                        return ast  # uncorrelated: eager scalar later
                    ssub_n[0] += 1
                    sfields = subplan.getRowType().getFieldList()
                    lhs_fields = plan.getRowType().getFieldList()
                    tmp = LogicalPlan(
                        "__combined__", [],
                        RelDataType(lhs_fields + sfields), None)
                    cond = None
                    for i, ka in enumerate(outer_keys):
                        eq = Call("=", [
                            self._resolve(ka, tmp),
                            InputRef(len(lhs_fields) + i,
                                     sfields[i].getType())],
                            SqlType("BOOLEAN"))
                        cond = eq if cond is None else Call(
                            "AND", [cond, eq], SqlType("BOOLEAN"))
                    plan = LogicalPlan(
                        "Join", [plan, subplan],
                        RelDataType(lhs_fields + sfields),
                        JoinNode("LEFT", cond))
                    return ("col", None, valname)
                return ast
            if ast[0] == "call":
                return (ast[0], ast[1], [_pull_ssubs(a) for a in ast[2]])
            if ast[0] == "cast":
                return (ast[0], _pull_ssubs(ast[1]), ast[2])
            if ast[0] == "case":
                return (ast[0],
                        [(_pull_ssubs(c), _pull_ssubs(v))
                         for c, v in ast[1]],
                        _pull_ssubs(ast[2]) if ast[2] is not None else None)
            if ast[0] == "agg":
                return (ast[0], ast[1], [_pull_ssubs(a) for a in ast[2]],
                        ast[3], ast[4])
            return ast

        stmt.items = [(_pull_ssubs(e), a) for e, a in stmt.items]
        for i, cj in enumerate(where_conjuncts):
            if not used[i]:
                where_conjuncts[i] = _pull_ssubs(cj)

        # EXISTS: equality-correlated → SEMI/ANTI join over the DISTINCT
        # correlation keys (DataFusion's decorrelation on the reference
        # side); uncorrelated → COUNT(*) scalar subquery > 0
        for substmt, negated in exists_subs:
            subplan, outer_keys = self._decorrelate_exists(substmt)
            if subplan is None:
                # uncorrelated EXISTS → scalar COUNT(*) comparison
                cnt = SelectStmt(
                    items=[(("agg", "count", [("star",)], False, None),
                            "c")],
                    from_tables=substmt.from_tables, joins=substmt.joins,
                    where=substmt.where)
                cmp_ast = ("call", "=" if negated else ">",
                           [("scalar_sub", cnt), ("lit", 0, "BIGINT")])
                cond = self._resolve(cmp_ast, plan)
                plan = LogicalPlan("Filter", [plan], plan.getRowType(),
                                   FilterNode(cond))
                continue
            plan = self._semi_anti_join(plan, subplan, outer_keys, negated)

        # IN-subquery joins: x IN (SELECT c FROM ...) ≡ SEMI join on x = c
        # over the DISTINCT subquery output; NOT IN ≡ ANTI (NULL-in-subquery
        # divergence documented in DESIGN.md). Equality-correlated subqueries
        # add their correlation keys to the join.
        for e_ast, substmt, negated in in_subs:
            if len(substmt.items) != 1:
                raise ValueError("IN subquery must select exactly one column")
            subplan, outer_keys = self._decorrelate_exists(
                substmt, lead_items=list(substmt.items))
            if subplan is None:
                subplan = self.build_stmt(substmt)
                outer_keys = []
            plan = self._semi_anti_join(plan, subplan,
                                        [e_ast] + outer_keys, negated,
                                        null_aware=negated)

        # leftover WHERE conjuncts → Filter (incl. scalar TRUE/FALSE)
        leftovers = [cj for i, cj in enumerate(where_conjuncts) if not used[i]
                     and not self._has_agg(cj)]
        if leftovers:
            ast = leftovers[0]
            for c in leftovers[1:]:
                ast = ("call", "AND", [ast, c])
            cond = self._resolve(ast, plan)
            plan = LogicalPlan("Filter", [plan], plan.getRowType(),
                               FilterNode(cond))

        # 2. expand stars
        items = []
        for e, alias in stmt.items:
            if e == ("star",):
                for f in plan.getRowType().getFieldList()[:n_user_fields]:
                    items.append((("col", f.qualifier, f.getName()), None))
            elif isinstance(e, tuple) and e[0] == "qstar":
                # t.* — the named table/alias's columns only
                q = e[1].lower()
                hit = False
                for f in plan.getRowType().getFieldList()[:n_user_fields]:
                    if (f.qualifier or "").lower() == q:
                        items.append((("col", f.qualifier, f.getName()),
                                      None))
                        hit = True
                if not hit:
                    raise KeyError(f"{e[1]}.*: no columns for that "
                                   "qualifier in scope")
            else:
                items.append((e, alias))

        # 2.5 window functions (reference rel/logical/window.py): compute
        # window columns over the joined/filtered input, rewrite select items
        win_asts = []

        def collect_windows(ast):
            if not isinstance(ast, tuple):
                return
            if ast[0] == "window":
                if ast not in win_asts:
                    win_asts.append(ast)
                return
            if ast[0] == "call":
                for a in ast[2]:
                    collect_windows(a)
            elif ast[0] == "cast":
                collect_windows(ast[1])
            elif ast[0] == "case":
                for cnd, v in ast[1]:
                    collect_windows(cnd)
                    collect_windows(v)
                if ast[2] is not None:
                    collect_windows(ast[2])

        for e, _ in items:
            collect_windows(e)
        if win_asts:
            if stmt.group_by or any(self._has_agg(e) for e, _ in items):
                raise NotImplementedError(
                    "window functions combined with GROUP BY/aggregates")
            plan, items = self._build_window(plan, items, win_asts)

        # 3. aggregate?
        has_agg = any(self._has_agg(e) for e, _ in items) or bool(
            stmt.group_by
        ) or (stmt.having is not None and self._has_agg(stmt.having))
        if has_agg:
            plan, items = self._build_aggregate(stmt, plan, items)
        # HAVING without aggregate context is just a filter
        elif stmt.having is not None:
            cond = self._resolve(stmt.having, plan)
            plan = LogicalPlan("Filter", [plan], plan.getRowType(),
                               FilterNode(cond))

        # 4. final projection (+ HIDDEN sort columns: ORDER BY may name
        # source columns that are not in the SELECT list — the reference
        # sorts before the final projection drops them (DataFusion plans
        # Sort over a projection containing the keys). We append them as
        # hidden output columns and strip them after Sort/Limit.)
        hidden = []
        if stmt.order_by and not has_agg and not stmt.distinct:
            def _matches_item(e):
                if e[0] == "lit" and isinstance(e[1], int):
                    return True
                for it, alias in items:
                    if it == e:
                        return True
                    if e[0] == "col":
                        name = e[2].lower()
                        if alias is not None and alias.lower() == name:
                            return True
                        if alias is None and it[0] == "col"                                 and it[2].lower() == name:
                            return True
                return False

            for e, _asc, _nf in stmt.order_by:
                if isinstance(e, tuple) and e[0] == "col"                         and not _matches_item(e)                         and all(e != h for h in hidden):
                    hidden.append(e)
        n_vis = len(items)
        if hidden:
            items = items + [(h, f"__sort_h{i}")
                             for i, h in enumerate(hidden)]
        plan = self._build_projection(plan, items)

        # 5. DISTINCT
        if stmt.distinct:
            fields = plan.getRowType().getFieldList()
            group_exprs = [InputRef(i, f.getType()) for i, f in enumerate(fields)]
            node = AggregateNode(group_exprs, [], distinct_node=True,
                                 distinct_columns=[f.getName() for f in fields])
            plan = LogicalPlan("Distinct", [plan], plan.getRowType(), node)

        # 6. ORDER BY / LIMIT (then strip any hidden sort columns —
        # Sort→Limit order keeps the device top-k fusion applicable)
        if stmt.order_by:
            keys = []
            for e, asc, nf in stmt.order_by:
                if e in hidden:
                    idx = n_vis + hidden.index(e)
                else:
                    idx = self._find_output(e, stmt, plan)
                keys.append((idx, asc, nf))
            plan = LogicalPlan("Sort", [plan], plan.getRowType(),
                               SortNode(keys))
        if stmt.limit is not None or stmt.offset:
            plan = LogicalPlan("Limit", [plan], plan.getRowType(),
                               LimitNode(stmt.limit, stmt.offset))
        if hidden:
            fields = plan.getRowType().getFieldList()[:n_vis]
            named = [(InputRef(i, f.getType()), f.getName())
                     for i, f in enumerate(fields)]
            plan = LogicalPlan("Projection", [plan], RelDataType(fields),
                               ProjectionNode(named))
        return plan

    # ------------------------------------------------------------ union
    def _build_union(self, u):
        """Left-fold of positional unions; non-ALL steps wrap in Distinct
        (reference: Union rel → dd.concat [+ drop_duplicates])."""
        from dask_sql_amd.planner.plan import UnionNode
        plan = self.build_stmt(u.branches[0])
        ops = u.ops or ["UNION"] * len(u.alls)
        for allf, op, br in zip(u.alls, ops, u.branches[1:]):
            rhs = self.build_stmt(br)
            lf = plan.getRowType().getFieldList()
            rf = rhs.getRowType().getFieldList()
            if len(lf) != len(rf):
                raise ValueError(f"{op} branches have different arity")
            if op in ("INTERSECT", "EXCEPT"):
                # distinct SEMI/ANTI join on all columns (the reference's
                # DataFusion rewrites Intersect/Except the same way;
                # NULL-key rows never match the equality join — documented
                # divergence from NULL-tolerant set semantics)
                node = AggregateNode(
                    [InputRef(i, f.getType()) for i, f in enumerate(rf)],
                    [], distinct_node=True,
                    distinct_columns=[f.getName() for f in rf])
                rhs = LogicalPlan("Distinct", [rhs], rhs.getRowType(), node)
                cond = None
                for i, (a, b) in enumerate(zip(lf, rf)):
                    eq = Call("=", [InputRef(i, a.getType()),
                                    InputRef(len(lf) + i, b.getType())],
                              SqlType("BOOLEAN"))
                    cond = eq if cond is None else Call(
                        "AND", [cond, eq], SqlType("BOOLEAN"))
                jt = "LEFTSEMI" if op == "INTERSECT" else "LEFTANTI"
                jnode = JoinNode(jt, cond)
                # DataFusion rewrites Intersect/Except with
                # null_equals_null=true: NULL keys compare equal
                jnode.null_equal = True
                plan = LogicalPlan("Join", [plan, rhs], RelDataType(lf),
                                   jnode)
                gexprs = [InputRef(i, f.getType()) for i, f in enumerate(lf)]
                node = AggregateNode(gexprs, [], distinct_node=True,
                                     distinct_columns=[f.getName()
                                                       for f in lf])
                plan = LogicalPlan("Distinct", [plan], plan.getRowType(),
                                   node)
                continue
            fields = []
            for a, b in zip(lf, rf):
                ta, tb = a.getType().getSqlType(), b.getType().getSqlType()
                ty = ta if ta == tb else _common_type(ta, tb)
                fields.append(Field(a.getName(), SqlType(ty)))
            plan = LogicalPlan("Union", [plan, rhs], RelDataType(fields),
                               UnionNode())
            if not allf:
                dfields = plan.getRowType().getFieldList()
                gexprs = [InputRef(i, f.getType())
                          for i, f in enumerate(dfields)]
                node = AggregateNode(gexprs, [], distinct_node=True,
                                     distinct_columns=[f.getName()
                                                       for f in dfields])
                plan = LogicalPlan("Distinct", [plan], plan.getRowType(),
                                   node)
        if u.order_by:
            names = [f.getName().lower()
                     for f in plan.getRowType().getFieldList()]
            keys = []
            for e, asc, nf in u.order_by:
                if e[0] == "col" and e[1] is None \
                        and e[2].lower() in names:
                    idx = names.index(e[2].lower())
                elif e[0] == "lit" and isinstance(e[1], int):
                    idx = e[1] - 1
                else:
                    raise NotImplementedError(
                        "ORDER BY over UNION must name an output column")
                keys.append((idx, asc, nf))
            plan = LogicalPlan("Sort", [plan], plan.getRowType(),
                               SortNode(keys))
        if u.limit is not None or u.offset:
            plan = LogicalPlan("Limit", [plan], plan.getRowType(),
                               LimitNode(u.limit, u.offset))
        return plan

    # ------------------------------------------------------------ window
    def _build_window(self, plan, items, win_asts):
        """Insert Projection(inputs + window arg/key exprs) → Window(specs)
        and rewrite select items to reference the window output columns
        (reference rel/logical/window.py:212-428)."""
        from dask_sql_amd.planner.plan import (ProjectionNode, WindowNode,
                                               WindowSpec)
        in_fields = plan.getRowType().getFieldList()
        pre_named = [(InputRef(i, f.getType()), f.getName())
                     for i, f in enumerate(in_fields)]
        pre_fields = [Field(f.getName(), f.getType(), qualifier=f.qualifier)
                      for f in in_fields]
        key_of = {}

        def idx_of(ast):
            key = repr(ast)
            if key in key_of:
                return key_of[key]
            e = self._resolve(ast, plan)
            if isinstance(e, InputRef):
                key_of[key] = e.getIndex()
                return e.getIndex()
            name = f"w_in{len(pre_named)}"
            pre_named.append((e, name))
            pre_fields.append(Field(name, SqlType(_expr_type(e))))
            key_of[key] = len(pre_named) - 1
            return key_of[key]

        specs = []
        win_out = {}
        for ast in win_asts:
            _, func, args, part, order, frame = ast
            if func == "single_value":
                func = "first_value"  # reference maps both to FirstValue
            arg_idx = None
            arg_t = None
            if args and args[0] != ("star",):
                arg_idx = idx_of(args[0])
                arg_t = pre_fields[arg_idx].getType().getSqlType()
            part_idx = [idx_of(p) for p in part]
            order_idx = [(idx_of(o[0]), o[1]) for o in order]
            order_nf = [o[2] if len(o) > 2 else None for o in order]
            if func in ("rank", "dense_rank", "lag", "lead") \
                    and not order_idx:
                raise ValueError(f"{func.upper()} requires ORDER BY in OVER")
            # ROW_NUMBER / FIRST_VALUE / LAST_VALUE without ORDER BY run in
            # input order (reference window.py row_number = range(1..n))
            offset, default = 1, None
            if func in ("lag", "lead"):

                def lit_of(a, what):
                    if a[0] == "lit":
                        return a[1]
                    if (a[0] == "call" and a[1] == "NEG"
                            and a[2][0][0] == "lit"):
                        return -a[2][0][1]
                    raise ValueError(f"LAG/LEAD {what} must be a literal")

                if len(args) > 1:
                    offset = int(lit_of(args[1], "offset"))
                if len(args) > 2:
                    default = lit_of(args[2], "default")
            if func in ("row_number", "rank", "dense_rank", "count"):
                ty = "BIGINT"
            elif func == "avg":
                ty = "DOUBLE"
            elif func == "sum":
                ty = "DOUBLE" if _is_float(arg_t or "BIGINT") else "BIGINT"
            else:  # min/max/lag/lead/first_value keep the arg type
                ty = arg_t or "BIGINT"
            if frame is not None:
                fk, lo, hi = frame
                if fk == "range":
                    raise NotImplementedError(
                        "RANGE frames with offsets (ROWS frames and the "
                        "default RANGE frame are supported)")
            name = f"w{len(specs)}__{func}"
            spec = WindowSpec(func, arg_idx, part_idx, order_idx,
                              name, SqlType(ty), offset=offset,
                              default=default, frame=frame)
            spec.order_nf = order_nf  # per-key NULLS FIRST/LAST (None =
            # engine default: nulls sort last)
            specs.append(spec)
            win_out[repr(ast)] = name
        if len(pre_named) > len(in_fields):
            plan = LogicalPlan("Projection", [plan], RelDataType(pre_fields),
                               ProjectionNode(pre_named))
        wfields = pre_fields + [Field(s.out_name, s.out_type) for s in specs]
        plan = LogicalPlan("Window", [plan], RelDataType(wfields),
                           WindowNode(specs))

        def rewrite(ast):
            if isinstance(ast, tuple):
                if ast[0] == "window":
                    return ("col", None, win_out[repr(ast)])
                if ast[0] == "call":
                    return ("call", ast[1], [rewrite(a) for a in ast[2]])
                if ast[0] == "cast":
                    return ("cast", rewrite(ast[1]), ast[2])
                if ast[0] == "case":
                    return ("case",
                            [(rewrite(c), rewrite(v)) for c, v in ast[1]],
                            rewrite(ast[2]) if ast[2] is not None else None)
            return ast

        items2 = []
        for e, alias in items:
            if alias is None and isinstance(e, tuple) and e[0] == "window":
                alias = f"{e[1].upper()}() OVER"
            items2.append((rewrite(e), alias))
        return plan, items2

    # ------------------------------------------------------------ aggregate
    def _build_aggregate(self, stmt, plan, items):
        # collect group exprs and agg calls (dedup by AST)
        group_asts = list(stmt.group_by)
        agg_asts = []

        def collect_aggs(ast):
            if not isinstance(ast, tuple):
                return
            if ast[0] == "agg":
                if ast not in agg_asts:
                    agg_asts.append(ast)
                return
            if ast[0] == "call":
                for a in ast[2]:
                    collect_aggs(a)
            elif ast[0] == "cast":
                collect_aggs(ast[1])
            elif ast[0] == "case":
                for c, v in ast[1]:
                    collect_aggs(c)
                    collect_aggs(v)
                if ast[2] is not None:
                    collect_aggs(ast[2])

        for e, _ in items:
            collect_aggs(e)
        if stmt.having is not None:
            collect_aggs(stmt.having)
        for e, _, _ in stmt.order_by:
            collect_aggs(e)

        # GROUP BY items may be aliases of select items or positions
        resolved_groups = []
        for g in group_asts:
            if g[0] == "lit" and isinstance(g[1], int):
                e, _ = items[g[1] - 1]
                resolved_groups.append(e)
            elif g[0] == "col" and g[1] is None:
                # alias of a select item?
                hit = None
                for e, alias in items:
                    if alias and alias.lower() == g[2].lower():
                        hit = e
                        break
                try:
                    self._resolve(g, plan)
                    hit = None  # real column wins
                except KeyError:
                    pass
                resolved_groups.append(hit if hit is not None else g)
            else:
                resolved_groups.append(g)

        # pre-projection: group exprs, agg args, agg filters
        pre_named = []  # (Expression, name)
        pre_fields = []
        name_of = {}

        def add_pre(ast, base_name):
            key = repr(ast)
            if key in name_of:
                return name_of[key]
            e = self._resolve(ast, plan)
            name = base_name
            k = 1
            existing = {n for _, n in pre_named}
            while name in existing:
                name = f"{base_name}_{k}"
                k += 1
            pre_named.append((e, name))
            src_q = None
            if ast[0] == "col":
                i = e.getIndex() if isinstance(e, InputRef) else None
                if i is not None:
                    src_q = plan.getRowType().getFieldList()[i].qualifier
            pre_fields.append(Field(name, SqlType(_expr_type(e)),
                                    qualifier=src_q))
            name_of[key] = len(pre_named) - 1
            return name_of[key]

        def base_name_for(ast):
            if ast[0] == "col":
                return ast[2]
            return f"expr{len(pre_named)}"

        group_idx = [add_pre(g, base_name_for(g)) for g in resolved_groups]

        agg_info = []  # (ast, func, arg_idx|None, filter_idx|None, distinct)
        for ast in agg_asts:
            _, func, args, distinct, filt = ast
            if len(args) == 1 and args[0] == ("star",):
                arg_idx = None  # COUNT(*)
            elif len(args) == 1:
                arg_idx = add_pre(args[0], base_name_for(args[0]))
            elif len(args) == 0:
                arg_idx = None
            else:
                raise NotImplementedError("multi-arg aggregates")
            filt_idx = add_pre(filt, f"filter{len(pre_named)}") \
                if filt is not None else None
            agg_info.append((ast, func, arg_idx, filt_idx, distinct))

        pre_plan = LogicalPlan("Projection", [plan], RelDataType(pre_fields),
                               ProjectionNode(pre_named))

        # aggregate node
        group_exprs = [InputRef(i, pre_fields[i].getType()) for i in group_idx]
        agg_calls = []
        out_fields = [pre_fields[i] for i in group_idx]
        agg_out_of = {}
        for ast, func, arg_idx, filt_idx, distinct in agg_info:
            if arg_idx is not None:
                args = [InputRef(arg_idx, pre_fields[arg_idx].getType())]
                arg_t = pre_fields[arg_idx].getType().getSqlType()
            else:
                args = []
                arg_t = "BIGINT"
            if func == "count":
                out_t = "BIGINT"
            elif func in ("every", "bool_and", "bool_or"):
                out_t = "BOOLEAN"
            elif func in ("avg", "stddev", "stddev_samp", "stddev_pop",
                          "var_samp", "var_pop", "variance"):
                out_t = "DOUBLE"
            elif func == "sum":
                out_t = "DOUBLE" if _is_float(arg_t) else "BIGINT"
            else:
                out_t = arg_t
            out_name = f"{func.upper()}({pre_fields[arg_idx].getName()})" \
                if arg_idx is not None else f"{func.upper()}(*)"
            k = 1
            existing = {f.getName() for f in out_fields}
            base = out_name
            while out_name in existing:
                out_name = f"{base}_{k}"
                k += 1
            filt_e = InputRef(filt_idx, pre_fields[filt_idx].getType()) \
                if filt_idx is not None else None
            agg_calls.append(AggCall(func, args, out_name, filt_e, distinct))
            out_fields.append(Field(out_name, SqlType(out_t)))
            agg_out_of[repr(ast)] = len(out_fields) - 1

        agg_node = AggregateNode(group_exprs, agg_calls)
        agg_plan = LogicalPlan("Aggregate", [pre_plan],
                               RelDataType(out_fields), agg_node)

        # rewrite select items over aggregate output
        group_out_of = {repr(g): gi for gi, g in enumerate(resolved_groups)}

        def rewrite(ast):
            if repr(ast) in group_out_of:
                i = group_out_of[repr(ast)]
                return ("col", None, out_fields[i].getName())
            if isinstance(ast, tuple) and ast[0] == "agg":
                i = agg_out_of[repr(ast)]
                return ("col", None, out_fields[i].getName())
            if isinstance(ast, tuple) and ast[0] == "call":
                return ("call", ast[1], [rewrite(a) for a in ast[2]])
            if isinstance(ast, tuple) and ast[0] == "cast":
                return ("cast", rewrite(ast[1]), ast[2])
            if isinstance(ast, tuple) and ast[0] == "case":
                return ("case",
                        [(rewrite(c), rewrite(v)) for c, v in ast[1]],
                        rewrite(ast[2]) if ast[2] is not None else None)
            return ast

        new_items = [(rewrite(e), alias) for e, alias in items]

        # HAVING → filter over aggregate output
        if stmt.having is not None:
            cond = self._resolve(rewrite(stmt.having), agg_plan)
            agg_plan = LogicalPlan("Filter", [agg_plan], agg_plan.getRowType(),
                                   FilterNode(cond))

        # stash rewrite for ORDER BY resolution
        self._agg_rewrite = rewrite
        return agg_plan, new_items

    # ----------------------------------------------------------- projection
    def _build_projection(self, plan, items):
        named = []
        fields = []
        # derive names, disambiguating duplicates with qualifiers
        # (context.py:890-898)
        base_names = []
        for e, alias in items:
            if alias:
                base_names.append(alias)
            elif e[0] == "col":
                base_names.append(e[2])
            elif e[0] == "agg":
                base_names.append(f"{e[1].upper()}()")
            else:
                base_names.append(None)
        counts = {}
        for n in base_names:
            if n is not None:
                counts[n.lower()] = counts.get(n.lower(), 0) + 1
        out_names = []
        for (e, alias), bn in zip(items, base_names):
            if alias:
                out_names.append(alias)
            elif bn is not None and counts.get(bn.lower(), 0) > 1 \
                    and e[0] == "col" and e[1] is not None:
                out_names.append(f"{e[1]}.{e[2]}")
            elif bn is not None:
                out_names.append(bn)
            else:
                out_names.append(f"EXPR${len(out_names)}")
        # a second pass: if still duplicated (both unqualified), qualify by
        # source field qualifier
        seen = {}
        for i, n in enumerate(out_names):
            seen.setdefault(n.lower(), []).append(i)
        for n, idxs in seen.items():
            if len(idxs) > 1:
                for i in idxs:
                    e, alias = items[i]
                    if alias is None and e[0] == "col":
                        ref = self._resolve(e, plan)
                        if isinstance(ref, InputRef):
                            f = plan.getRowType().getFieldList()[ref.getIndex()]
                            if f.qualifier:
                                out_names[i] = f"{f.qualifier}.{f.getName()}"

        for (e, alias), name in zip(items, out_names):
            expr = self._resolve(e, plan)
            named.append((expr, name))
            src_q = None
            if isinstance(expr, InputRef):
                src_q = plan.getRowType().getFieldList()[expr.getIndex()].qualifier
            fields.append(Field(name, SqlType(_expr_type(expr)),
                                qualifier=src_q))
        return LogicalPlan("Projection", [plan], RelDataType(fields),
                           ProjectionNode(named))

    def _find_output(self, ast, stmt, plan) -> int:
        fields = plan.getRowType().getFieldList()
        orig = ast
        if ast[0] == "lit" and isinstance(ast[1], int):
            return ast[1] - 1
        if hasattr(self, "_agg_rewrite"):
            ast = self._agg_rewrite(ast)
        if ast[0] == "col" and ast[1] is None:
            for i, f in enumerate(fields):
                if f.getName().lower() == ast[2].lower():
                    return i
        if ast[0] == "col" and ast[1] is not None:
            # qualified ORDER BY key (ORDER BY t.c) against the output:
            # match name + qualifier, else the name alone when unique
            q, n = ast[1].lower(), ast[2].lower()
            hits = [i for i, f in enumerate(fields)
                    if f.getName().lower() == n
                    and (f.qualifier or "").lower() == q]
            if not hits:
                hits = [i for i, f in enumerate(fields)
                        if f.getName().lower() == n]
            if len(hits) == 1:
                return hits[0]
        # structural match against select items (ORDER BY SUM(v) where the
        # same aggregate appears in the SELECT list, possibly aliased)
        for i, (e, alias) in enumerate(stmt.items):
            if e == orig or e == ast:
                return i
        raise KeyError(f"ORDER BY expression not in output: {ast!r} "
                       "(an ORDER BY aggregate must also appear in SELECT)")
