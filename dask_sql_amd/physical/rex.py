"""Rex → VM-program compiler.

Replaces the reference RexConverter dispatch + RexCallPlugin evaluation
(physical/rex/convert.py:47-76, rex/core/call.py:1047-1156): instead of
per-operator pandas calls, an Expression tree compiles into one typed postfix
program (include/dsxhip.h DsxOp) interpreted per row inside the HIP kernels.
"""
from __future__ import annotations

from dask_sql_amd.planner.plan import Call, Expression, InputRef, Literal
from dask_sql_amd import runtime as rt

# opcodes (include/dsxhip.h)
OP_COL, OP_LIT_F64, OP_LIT_I64, OP_LIT_NULL = 1, 2, 3, 4
OP_ADD_F64, OP_SUB_F64, OP_MUL_F64, OP_DIV_F64 = 10, 11, 12, 13
OP_ADD_I64, OP_SUB_I64, OP_MUL_I64, OP_DIV_I64, OP_MOD_I64 = 14, 15, 16, 17, 18
OP_LT_F64, OP_LE_F64, OP_GT_F64, OP_GE_F64, OP_EQ_F64, OP_NE_F64 = (
    20, 21, 22, 23, 24, 25)
OP_LT_I64, OP_LE_I64, OP_GT_I64, OP_GE_I64, OP_EQ_I64, OP_NE_I64 = (
    30, 31, 32, 33, 34, 35)
OP_AND, OP_OR, OP_NOT, OP_IS_NULL, OP_IS_NOT_NULL = 40, 41, 42, 43, 44
OP_I64_TO_F64, OP_F64_TO_I64, OP_BITS_F64 = 50, 51, 52
OP_SELECT, OP_NEG_F64, OP_NEG_I64, OP_SQRT_F64 = 60, 61, 62, 63
OP_ABS_I64, OP_ABS_F64, OP_FLOOR_F64, OP_CEIL_F64 = 64, 65, 66, 67
OP_RINT_F64, OP_EXP_F64, OP_LN_F64, OP_POW_F64 = 68, 69, 70, 71
OP_YEAR, OP_MONTH, OP_DAY = 72, 73, 74
OP_FLOORMOD_I64 = 75
OP_SIN_F64, OP_COS_F64, OP_TAN_F64 = 76, 77, 78
OP_ASIN_F64, OP_ACOS_F64, OP_ATAN_F64, OP_ATAN2_F64 = 79, 80, 81, 82

# VM value kinds
KI, KF, KB = "i", "f", "b"  # int64-like, float64, boolean

_SQL_TO_KIND = {
    "BIGINT": KI, "INTEGER": KI, "SMALLINT": KI, "TINYINT": KI, "DATE": KI,
    "TIMESTAMP": KI, "VARCHAR": KI,  # dict codes
    "DOUBLE": KF, "FLOAT": KF, "DECIMAL": KF,
    "BOOLEAN": KB, "NULL": KI,
}

_DSX_KIND = {rt.I64: KI, rt.I32: KI, rt.I8: KI, rt.BOOL8: KB,
             rt.F64: KF, rt.F32: KF}

_CMP = {"=": (OP_EQ_I64, OP_EQ_F64), "<>": (OP_NE_I64, OP_NE_F64),
        "<": (OP_LT_I64, OP_LT_F64), "<=": (OP_LE_I64, OP_LE_F64),
        ">": (OP_GT_I64, OP_GT_F64), ">=": (OP_GE_I64, OP_GE_F64)}
_ARITH = {"+": (OP_ADD_I64, OP_ADD_F64), "-": (OP_SUB_I64, OP_SUB_F64),
          "*": (OP_MUL_I64, OP_MUL_F64), "/": (OP_DIV_I64, OP_DIV_F64)}


class RexCompileError(NotImplementedError):
    pass


def _lit_int(o):
    if isinstance(o, Literal):
        try:
            return int(o.getValue())
        except (TypeError, ValueError):
            return None
    if isinstance(o, Call) and o.getOperatorName() == "NEG" \
            and isinstance(o.getOperands()[0], Literal):
        return -int(o.getOperands()[0].getValue())
    return None


def dict_string_fn(expr, dicts):
    """If expr is a chain of string functions (UPPER/LOWER/SUBSTRING) over a
    dict-encoded column, return (col_index, python_fn) computing the string
    result per dictionary entry — the whole transform runs once over the
    (small, host-resident) dictionary, never per row (reference
    rex/core/call.py:1069-1135 string operations). None if not such a
    chain."""
    if isinstance(expr, InputRef):
        i = expr.getIndex()
        if i < len(dicts) and dicts[i] is not None:
            return i, (lambda s: s)
        return None
    if isinstance(expr, Call):
        op = expr.getOperatorName().upper()
        ops_ = expr.getOperands()
        sub = dict_string_fn(ops_[0], dicts) if ops_ else None
        if sub is None:
            return None
        i, f = sub
        if op in ("UPPER", "LOWER") and len(ops_) == 1:
            if op == "UPPER":
                return i, (lambda s, f=f: f(s).upper())
            return i, (lambda s, f=f: f(s).lower())
        if op in ("SUBSTRING", "SUBSTR") and len(ops_) in (2, 3):
            a = _lit_int(ops_[1])
            ln = _lit_int(ops_[2]) if len(ops_) == 3 else None
            if a is None or (len(ops_) == 3 and ln is None):
                return None
            if ln is not None and ln < 0:
                return None
            # Calcite: FROM 0/negative counts from the start without
            # shifting the window (test_rex.py:614 SUBSTRING(a FROM -1)
            # yields the whole string)
            start = max(a - 1, 0)

            def g(s, f=f, start=start, ln=ln):
                t = f(s)
                return t[start:start + ln] if ln is not None else t[start:]

            return i, g
        if op == "INITCAP" and len(ops_) == 1:
            return i, (lambda s, f=f: f(s).title())
        if op == "OVERLAY" and len(ops_) in (3, 4):
            # OVERLAY(x PLACING y FROM n [FOR m]) — reference
            # rex/core/call.py OverlayOperation (1-based start; start<=0
            # clamps to 0; splice length defaults to len(y))
            if not (isinstance(ops_[1], Literal)
                    and isinstance(ops_[1].getValue(), str)):
                return None
            start = _lit_int(ops_[2])
            if start is None:
                return None
            length = _lit_int(ops_[3]) if len(ops_) == 4 else None
            if len(ops_) == 4 and length is None:
                return None
            repl = ops_[1].getValue()

            def ov(s, f=f, repl=repl, start=start, length=length):
                t = f(s)
                st = 0 if start <= 0 else start - 1
                ln = len(repl) if length is None else length
                return t[:st] + repl + t[st + ln:]

            return i, ov
        if op == "TRIM" and len(ops_) == 3:
            if not all(isinstance(o, Literal) for o in ops_[1:]):
                return None
            mode = str(ops_[1].getValue()).upper()
            ch = str(ops_[2].getValue())

            def g(s, f=f, mode=mode, ch=ch):
                t = f(s)
                if mode == "LEADING":
                    return t.lstrip(ch)
                if mode == "TRAILING":
                    return t.rstrip(ch)
                return t.strip(ch)

            return i, g
        if op == "REPLACE" and len(ops_) == 3:
            if not all(isinstance(o, Literal) for o in ops_[1:]):
                return None
            s1 = str(ops_[1].getValue())
            s2 = str(ops_[2].getValue())
            return i, (lambda s, f=f, s1=s1, s2=s2: f(s).replace(s1, s2))
        if op == "CONCAT":
            parts = []
            ci = None
            for o in ops_:
                if isinstance(o, Literal) and isinstance(o.getValue(), str):
                    parts.append(("lit", o.getValue()))
                    continue
                sub2 = dict_string_fn(o, dicts)
                if sub2 is None:
                    return None
                if ci is None:
                    ci = sub2[0]
                elif sub2[0] != ci:
                    return None  # concat across two dict columns: per-row
                parts.append(("fn", sub2[1]))
            if ci is None:
                return None

            def g(s, parts=parts):
                return "".join(p[1] if p[0] == "lit" else p[1](s)
                               for p in parts)

            return ci, g
    return None


def fold_string_literal(expr):
    """Evaluate a literal-only string-function chain to a constant str
    (REPLACE('Another String', ...) etc. — test_rex.py:624)."""
    if isinstance(expr, Literal) and isinstance(expr.getValue(), str):
        return expr.getValue()
    if not isinstance(expr, Call):
        return None
    op = expr.getOperatorName().upper()
    ops_ = expr.getOperands()
    vals = [fold_string_literal(o) if i == 0 or op == "CONCAT"
            else (o.getValue() if isinstance(o, Literal) else _lit_int(o))
            for i, o in enumerate(ops_)]
    if any(v is None for v in vals):
        return None
    s = vals[0]
    if op == "UPPER":
        return s.upper()
    if op == "LOWER":
        return s.lower()
    if op == "INITCAP":
        return s.title()
    if op == "REPLACE":
        return s.replace(str(vals[1]), str(vals[2]))
    if op == "TRIM":
        mode, ch = str(vals[1]).upper(), str(vals[2])
        return s.lstrip(ch) if mode == "LEADING" else \
            s.rstrip(ch) if mode == "TRAILING" else s.strip(ch)
    if op in ("SUBSTRING", "SUBSTR"):
        a0 = _lit_int(ops_[1])
        if a0 is None:
            return None
        a = max(a0 - 1, 0)
        ln = _lit_int(ops_[2]) if len(ops_) > 2 else None
        return s[a:a + ln] if ln is not None else s[a:]
    if op == "CONCAT":
        return "".join(str(v) for v in vals)
    return None


def dict_int_fn(expr, dicts):
    """String function with an INTEGER result over a dict column
    (CHAR_LENGTH): returns (col_index, per-entry int fn) — evaluated once
    over the dictionary, applied per row via a LUT gather."""
    if not isinstance(expr, Call):
        return None
    op = expr.getOperatorName().upper()
    if op in ("CHAR_LENGTH", "CHARACTER_LENGTH", "LENGTH") \
            and len(expr.getOperands()) == 1:
        sub = dict_string_fn(expr.getOperands()[0], dicts)
        if sub is None:
            return None
        i, f = sub
        return i, (lambda s, f=f: len(f(s)))
    if op == "TO_TIMESTAMP" and len(expr.getOperands()) == 2:
        # string → timestamp: strptime over the dictionary (reference
        # rex/core/call.py ToTimestampOperation string case)
        ops_ = expr.getOperands()
        sub = dict_string_fn(ops_[0], dicts)
        if sub is None or not isinstance(ops_[1], Literal):
            return None
        fmt = str(ops_[1].getValue())
        i, f = sub

        def ts_of(s, f=f, fmt=fmt):
            import numpy as _np
            from datetime import datetime as _dt
            return int(_np.datetime64(_dt.strptime(f(s), fmt),
                                      "ns").astype("int64"))

        return i, ts_of
    if op == "POSITION" and len(expr.getOperands()) in (2, 3):
        # POSITION(needle IN hay [FROM start]) — 1-based, 0 when absent
        # (reference rex/core/call.py PositionOperation)
        ops_ = expr.getOperands()
        sub = dict_string_fn(ops_[0], dicts)
        if sub is None or not (isinstance(ops_[1], Literal)
                               and isinstance(ops_[1].getValue(), str)):
            return None
        start = _lit_int(ops_[2]) if len(ops_) > 2 else 1
        if start is None:
            return None
        i, f = sub
        needle = ops_[1].getValue()
        return i, (lambda s, f=f, n=needle, st=start:
                   f(s).find(n, max(st - 1, 0)) + 1)
    return None


def _like_regex(pattern: str, escape=None, mode="LIKE"):
    """SQL LIKE/ILIKE/SIMILAR pattern → compiled regex. LIKE: % = any run,
    _ = any single char, everything else literal; an ESCAPE char makes the
    following char literal. ILIKE adds IGNORECASE. SIMILAR TO additionally
    passes the SQL:1999 regex metacharacters ( ) [ ] | * + ? { } through
    (reference rex/core/call.py LIKE/SIMILAR lowering)."""
    import re
    similar_meta = set("()[]|*+?{}-^,")
    out = []
    i = 0
    while i < len(pattern):
        ch = pattern[i]
        if escape and ch == escape and i + 1 < len(pattern):
            out.append(re.escape(pattern[i + 1]))
            i += 2
            continue
        if ch == "%":
            out.append(".*")
        elif ch == "_":
            out.append(".")
        elif mode == "SIMILAR" and ch in similar_meta:
            out.append(ch)
        else:
            out.append(re.escape(ch))
        i += 1
    flags = re.DOTALL | (re.IGNORECASE if mode == "ILIKE" else 0)
    return re.compile("".join(out), flags)


class RexCompiler:
    """Compiles an Expression over a column scope.

    cols: list of DeviceColumn in frontend order (InputRef index order);
    dictionaries: optional per-column list of string dictionaries (for
    dict-encoded VARCHAR equality — SURVEY §8f2 dict-code compare).
    """

    def __init__(self, cols, dictionaries=None):
        self.cols = cols
        self.dicts = dictionaries or [None] * len(cols)
        self.prog: list[tuple] = []

    # ---- emit helpers -----------------------------------------------------
    def _emit(self, op, arg0=0, imm=0):
        self.prog.append((op, arg0, imm))

    def _to_f(self, kind):
        if kind == KF:
            return KF
        self._emit(OP_I64_TO_F64)
        return KF

    # ---- main -------------------------------------------------------------
    def compile(self, expr: Expression) -> str:
        """Appends code; returns result kind."""
        if isinstance(expr, InputRef):
            i = expr.getIndex()
            self._emit(OP_COL, i)
            return _DSX_KIND[self.cols[i].dtype]
        if isinstance(expr, Literal):
            v = expr.getValue()
            if isinstance(v, tuple):
                raise RexCompileError("unfolded INTERVAL literal")
            if v is None:
                self._emit(OP_LIT_NULL)
                return KI
            t = expr.getType().getSqlType() if expr.getType() else None
            if isinstance(v, bool):
                self._emit(OP_LIT_I64, 0, 1 if v else 0)
                return KB
            if isinstance(v, float):
                self._emit(OP_LIT_F64, 0, float(v))
                return KF
            if isinstance(v, str):
                raise RexCompileError(
                    "string literal outside dict-compare context")
            self._emit(OP_LIT_I64, 0, int(v))
            return KI
        if isinstance(expr, Call):
            return self._compile_call(expr)
        raise RexCompileError(f"cannot compile {expr!r}")

    def _compile_call(self, call: Call) -> str:
        op = call.getOperatorName()
        ops = call.getOperands()
        if op in _CMP:
            return self._compile_cmp(op, ops)
        if op in _ARITH:
            a, b = ops
            # pre-scan kinds to decide int vs float path
            ka = self._peek_kind(a)
            kb = self._peek_kind(b)
            if KF in (ka, kb):
                k = self.compile(a)
                self._to_f(k)
                k = self.compile(b)
                self._to_f(k)
                self._emit(_ARITH[op][1])
                return KF
            self.compile(a)
            self.compile(b)
            self._emit(_ARITH[op][0])
            return KI
        if op == "AND":
            self.compile(ops[0])
            self.compile(ops[1])
            self._emit(OP_AND)
            return KB
        if op == "OR":
            self.compile(ops[0])
            self.compile(ops[1])
            self._emit(OP_OR)
            return KB
        if op == "NOT":
            self.compile(ops[0])
            self._emit(OP_NOT)
            return KB
        if op == "IS NULL":
            self.compile(ops[0])
            self._emit(OP_IS_NULL)
            return KB
        if op == "IS NOT NULL":
            self.compile(ops[0])
            self._emit(OP_IS_NOT_NULL)
            return KB
        if op == "NEG":
            k = self.compile(ops[0])
            self._emit(OP_NEG_F64 if k == KF else OP_NEG_I64)
            return k
        if op == "CAST":
            target = call.getType().getSqlType()
            k = self.compile(ops[0])
            tk = _SQL_TO_KIND.get(target, KF)
            if tk == KF and k != KF:
                self._emit(OP_I64_TO_F64)
                return KF
            if tk in (KI, KB) and k == KF:
                self._emit(OP_F64_TO_I64)  # trunc (mappings.py:346-353)
                return KI
            return k
        if op == "CASE":
            # operands: cond1, val1, cond2, val2, ..., else
            return self._compile_case(ops)
        if op in ("LIKE", "ILIKE", "SIMILAR"):
            return self._compile_like(ops, op)
        if op == "COALESCE":
            # right-fold of SELECT(arg IS NOT NULL, arg, rest)
            # (rex/core/call.py CoalesceOperation)
            target = KF if any(self._peek_kind(o) == KF for o in ops) \
                else self._peek_kind(ops[-1])

            def emit(i):
                if i == len(ops) - 1:
                    k = self.compile(ops[i])
                    if target == KF:
                        self._to_f(k)
                    return
                self.compile(ops[i])
                self._emit(OP_IS_NOT_NULL)
                k = self.compile(ops[i])
                if target == KF:
                    self._to_f(k)
                emit(i + 1)
                self._emit(OP_SELECT)

            emit(0)
            return target
        if op == "NULLIF":
            # SELECT(a = b, NULL, a); NULL condition falls through to a —
            # exactly SQL NULLIF (rex/core/call.py NullIf)
            a, b = ops
            ka, kb = self._peek_kind(a), self._peek_kind(b)
            use_f = KF in (ka, kb)
            k = self.compile(a)
            if use_f:
                self._to_f(k)
            k = self.compile(b)
            if use_f:
                self._to_f(k)
            self._emit(OP_EQ_F64 if use_f else OP_EQ_I64)
            self._emit(OP_LIT_NULL)
            k = self.compile(a)
            self._emit(OP_SELECT)
            return k
        # scalar math + date extraction (rex/core/call.py scalar operations:
        # abs/floor/ceil/round via numpy, exp/log/power, year/month/day)
        if op == "ABS":
            k = self.compile(ops[0])
            self._emit(OP_ABS_F64 if k == KF else OP_ABS_I64)
            return k
        _f1 = {"FLOOR": OP_FLOOR_F64, "CEIL": OP_CEIL_F64, "CEILING":
               OP_CEIL_F64, "EXP": OP_EXP_F64, "LN": OP_LN_F64,
               "LOG": OP_LN_F64, "SQRT": OP_SQRT_F64,
               "SIN": OP_SIN_F64, "COS": OP_COS_F64, "TAN": OP_TAN_F64,
               "ASIN": OP_ASIN_F64, "ACOS": OP_ACOS_F64,
               "ATAN": OP_ATAN_F64}
        if op in _f1:
            self._to_f(self.compile(ops[0]))
            self._emit(_f1[op])
            return KF
        if op == "ROUND":
            self._to_f(self.compile(ops[0]))
            if len(ops) == 2:
                if not isinstance(ops[1], Literal):
                    raise RexCompileError("ROUND digits must be a literal")
                p = 10.0 ** int(ops[1].getValue())
                self._emit(OP_LIT_F64, 0, p)
                self._emit(OP_MUL_F64)
                self._emit(OP_RINT_F64)  # ties-to-even like numpy round
                self._emit(OP_LIT_F64, 0, p)
                self._emit(OP_DIV_F64)
            else:
                self._emit(OP_RINT_F64)
            return KF
        if op in ("POWER", "POW"):
            self._to_f(self.compile(ops[0]))
            self._to_f(self.compile(ops[1]))
            self._emit(OP_POW_F64)
            return KF
        if op == "ATAN2":
            self._to_f(self.compile(ops[0]))
            self._to_f(self.compile(ops[1]))
            self._emit(OP_ATAN2_F64)
            return KF
        if op == "COT":
            # reference: da.cos/da.sin ratio; 1/tan matches to f64 ulp
            # except exactly at poles, where both are huge finite values
            self._emit(OP_LIT_F64, 0, 1.0)
            self._to_f(self.compile(ops[0]))
            self._emit(OP_TAN_F64)
            self._emit(OP_DIV_F64)
            return KF
        if op == "MOD":
            # reference evaluates operator.mod on pandas = FLOOR-mod
            # (MOD(-5,3) = 1), not C truncated remainder.
            # Floats: a - floor(a/b)*b (python float % semantics; b=0
            # yields NaN like pandas)
            if KF in (self._peek_kind(ops[0]), self._peek_kind(ops[1])):
                self._to_f(self.compile(ops[0]))
                self._to_f(self.compile(ops[0]))
                self._to_f(self.compile(ops[1]))
                self._emit(OP_DIV_F64)
                self._emit(OP_FLOOR_F64)
                self._to_f(self.compile(ops[1]))
                self._emit(OP_MUL_F64)
                self._emit(OP_SUB_F64)
                return KF
            self.compile(ops[0])
            self.compile(ops[1])
            self._emit(OP_FLOORMOD_I64)
            return KI
        _dx = {"EXTRACT_YEAR": OP_YEAR, "YEAR": OP_YEAR,
               "EXTRACT_MONTH": OP_MONTH, "MONTH": OP_MONTH,
               "EXTRACT_DAY": OP_DAY, "DAY": OP_DAY,
               "DAYOFMONTH": OP_DAY}
        _tx = {"EXTRACT_HOUR": (3_600_000_000_000, 24),
               "HOUR": (3_600_000_000_000, 24),
               "EXTRACT_MINUTE": (60_000_000_000, 60),
               "MINUTE": (60_000_000_000, 60),
               "EXTRACT_SECOND": (1_000_000_000, 60),
               "SECOND": (1_000_000_000, 60)}
        if op in _dx or op in _tx:
            t0 = getattr(ops[0], "getType", lambda: None)()
            is_ts = t0 is not None and t0.getSqlType() == "TIMESTAMP"
            k = self.compile(ops[0])
            if k != KI:
                raise RexCompileError(f"{op} needs a DATE/TIMESTAMP operand")
            if op in _tx:
                if not is_ts:
                    raise RexCompileError(f"{op} needs a TIMESTAMP operand")
                unit_ns, modulus = _tx[op]
                self._emit(OP_LIT_I64, 0, unit_ns)
                self._emit(OP_DIV_I64)
                self._emit(OP_LIT_I64, 0, modulus)
                self._emit(OP_MOD_I64)
                return KI
            if is_ts:
                # ns → days (trunc; pre-1970 sub-day values shift a day —
                # documented)
                self._emit(OP_LIT_I64, 0, 86_400_000_000_000)
                self._emit(OP_DIV_I64)
            self._emit(_dx[op])
            return KI
        if op == "EXTRACT_DATE" or op.startswith("FLOOR_TO_") \
                or op.startswith("CEIL_TO_"):
            return self._compile_dt_trunc(op, ops)
        if op == "LAST_DAY":
            return self._compile_last_day(ops)
        if op.startswith("EXTRACT_"):
            return self._compile_extract_ext(op, ops)
        raise RexCompileError(f"operator {op} not supported on GPU path")

    _DAY_NS = 86_400_000_000_000

    def _emit_days(self, ast, is_ts):
        """Push the operand's floor-div days-since-epoch (exact for
        pre-1970: x - floormod(x, day) is a day multiple, so trunc div is
        floor div)."""
        k = self.compile(ast)
        if k != KI:
            raise RexCompileError("datetime op needs DATE/TIMESTAMP")
        if is_ts:
            self.compile(ast)
            self._emit(OP_LIT_I64, 0, self._DAY_NS)
            self._emit(OP_FLOORMOD_I64)
            self._emit(OP_SUB_I64)
            self._emit(OP_LIT_I64, 0, self._DAY_NS)
            self._emit(OP_DIV_I64)

    def _compile_dt_trunc(self, op, ops):
        """FLOOR/CEIL(x TO unit) and EXTRACT(DATE FROM x) on the VM's
        existing integer ops (reference rex/core/call.py CeilFloorDatetime
        / ExtractOperation date truncation). Month/year starts come from
        the epoch-day civil-calendar ops (OP_DAY / OP_YEAR) plus the
        Gregorian leap-count identity; all intermediate divisions see
        positive years, so trunc division is floor division."""
        t0 = getattr(ops[0], "getType", lambda: None)()
        is_ts = t0 is not None and t0.getSqlType() == "TIMESTAMP"
        x = ops[0]
        sub_ns = {"HOUR": 3_600_000_000_000, "MINUTE": 60_000_000_000,
                  "SECOND": 1_000_000_000, "MILLISECOND": 1_000_000,
                  "MICROSECOND": 1_000}
        if op == "EXTRACT_DATE":
            self._emit_days(x, is_ts)
            return KI
        kind, unit = op.split("_TO_")
        if unit == "DAY" and not is_ts:
            k = self.compile(x)  # a DATE is already day-aligned
            if k != KI:
                raise RexCompileError("FLOOR TO needs DATE/TIMESTAMP")
            return KI
        if unit in sub_ns or (unit == "DAY" and is_ts):
            if not is_ts:
                raise RexCompileError(f"{op} needs a TIMESTAMP operand")
            u = self._DAY_NS if unit == "DAY" else sub_ns[unit]
            if kind == "FLOOR":
                self.compile(x)
                self.compile(x)
                self._emit(OP_LIT_I64, 0, u)
                self._emit(OP_FLOORMOD_I64)
                self._emit(OP_SUB_I64)
            else:  # CEIL: x + floormod(-x, u)
                self.compile(x)
                self._emit(OP_LIT_I64, 0, 0)
                self.compile(x)
                self._emit(OP_SUB_I64)
                self._emit(OP_LIT_I64, 0, u)
                self._emit(OP_FLOORMOD_I64)
                self._emit(OP_ADD_I64)
            return KI
        if kind == "CEIL" and unit in ("MONTH", "YEAR"):
            # CEIL(x) = start of the period FOLLOWING (x - 1 tick): a
            # boundary x decrements into the previous period (whose next
            # start is x itself), any other x stays in its own period.
            # Avoids a SELECT and its double re-emission — the stack VM
            # has no DUP, every reuse re-emits its subsequence, and the
            # program must stay inside DSX_MAX_PROG.
            def emit_pred_days():
                # day count of (x - 1ns) / (x - 1 day)
                def v():
                    k = self.compile(x)
                    if k != KI:
                        raise RexCompileError(
                            "CEIL TO needs DATE/TIMESTAMP")
                    self._emit(OP_LIT_I64, 0, 1)
                    self._emit(OP_SUB_I64)

                if not is_ts:
                    v()
                    return
                v()
                v()
                self._emit(OP_LIT_I64, 0, self._DAY_NS)
                self._emit(OP_FLOORMOD_I64)
                self._emit(OP_SUB_I64)
                self._emit(OP_LIT_I64, 0, self._DAY_NS)
                self._emit(OP_DIV_I64)

            if unit == "MONTH":
                def bumped():
                    # month start of (x-1) plus 31 — always inside (or on
                    # the first day of) the FOLLOWING month
                    emit_pred_days()
                    emit_pred_days()
                    self._emit(OP_DAY)
                    self._emit(OP_LIT_I64, 0, 1)
                    self._emit(OP_SUB_I64)
                    self._emit(OP_SUB_I64)
                    self._emit(OP_LIT_I64, 0, 31)
                    self._emit(OP_ADD_I64)

                bumped()
                bumped()
                self._emit(OP_DAY)
                self._emit(OP_LIT_I64, 0, 1)
                self._emit(OP_SUB_I64)
                self._emit(OP_SUB_I64)
            else:
                # jan1_days(Y) with Y = YEAR(x-1) + 1 — the same Gregorian
                # identity FLOOR_TO_YEAR uses, evaluated for the next year
                def emit_y1():
                    emit_pred_days()
                    self._emit(OP_YEAR)
                    self._emit(OP_LIT_I64, 0, 1)
                    self._emit(OP_ADD_I64)

                def leap1(div):
                    emit_y1()
                    self._emit(OP_LIT_I64, 0, 1)
                    self._emit(OP_SUB_I64)
                    self._emit(OP_LIT_I64, 0, div)
                    self._emit(OP_DIV_I64)

                emit_y1()
                self._emit(OP_LIT_I64, 0, 365)
                self._emit(OP_MUL_I64)
                leap1(4)
                self._emit(OP_ADD_I64)
                leap1(100)
                self._emit(OP_SUB_I64)
                leap1(400)
                self._emit(OP_ADD_I64)
                self._emit(OP_LIT_I64, 0, 719_527)
                self._emit(OP_SUB_I64)
            if is_ts:
                self._emit(OP_LIT_I64, 0, self._DAY_NS)
                self._emit(OP_MUL_I64)
            return KI
        if unit == "MONTH":
            # month_start_days = days - (dayofmonth(days) - 1)
            self._emit_days(x, is_ts)
            self._emit_days(x, is_ts)
            self._emit(OP_DAY)
            self._emit(OP_LIT_I64, 0, 1)
            self._emit(OP_SUB_I64)
            self._emit(OP_SUB_I64)
        elif unit == "YEAR":
            # jan1_days = 365*y + (y-1)/4 - (y-1)/100 + (y-1)/400 - 719527
            def emit_y():
                self._emit_days(x, is_ts)
                self._emit(OP_YEAR)

            def emit_leap_term(div):
                emit_y()
                self._emit(OP_LIT_I64, 0, 1)
                self._emit(OP_SUB_I64)
                self._emit(OP_LIT_I64, 0, div)
                self._emit(OP_DIV_I64)

            emit_y()
            self._emit(OP_LIT_I64, 0, 365)
            self._emit(OP_MUL_I64)
            emit_leap_term(4)
            self._emit(OP_ADD_I64)
            emit_leap_term(100)
            self._emit(OP_SUB_I64)
            emit_leap_term(400)
            self._emit(OP_ADD_I64)
            self._emit(OP_LIT_I64, 0, 719_527)
            self._emit(OP_SUB_I64)
        else:
            raise RexCompileError(f"FLOOR TO {unit} not supported")
        if is_ts:
            self._emit(OP_LIT_I64, 0, self._DAY_NS)
            self._emit(OP_MUL_I64)
        return KI

    def _compile_last_day(self, ops):
        """LAST_DAY(x) = x + MonthEnd(1) (reference call.py last_day):
        month end of x, ROLLING to the next month's end when x is already
        on one (pandas anchor semantics). Day identity:
        ld_days = MF(MF(days+1) + 31) - 1 with MF(g) = g - (DOM(g)-1);
        time of day is preserved."""
        t0 = getattr(ops[0], "getType", lambda: None)()
        is_ts = t0 is not None and t0.getSqlType() == "TIMESTAMP"
        x = ops[0]

        def emit_days1():
            # day count of x, plus one
            self._emit_days(x, is_ts)
            self._emit(OP_LIT_I64, 0, 1)
            self._emit(OP_ADD_I64)

        def emit_mf_days1():
            # month start of (days+1)
            emit_days1()
            emit_days1()
            self._emit(OP_DAY)
            self._emit(OP_LIT_I64, 0, 1)
            self._emit(OP_SUB_I64)
            self._emit(OP_SUB_I64)

        def emit_bumped():
            emit_mf_days1()
            self._emit(OP_LIT_I64, 0, 31)
            self._emit(OP_ADD_I64)

        if is_ts:
            # time of day rides on top of the day result
            k = self.compile(x)
            if k != KI:
                raise RexCompileError("LAST_DAY needs DATE/TIMESTAMP")
            self._emit(OP_LIT_I64, 0, self._DAY_NS)
            self._emit(OP_FLOORMOD_I64)
        emit_bumped()
        emit_bumped()
        self._emit(OP_DAY)
        self._emit(OP_LIT_I64, 0, 1)
        self._emit(OP_SUB_I64)
        self._emit(OP_SUB_I64)
        self._emit(OP_LIT_I64, 0, 1)
        self._emit(OP_SUB_I64)
        if is_ts:
            self._emit(OP_LIT_I64, 0, self._DAY_NS)
            self._emit(OP_MUL_I64)
            self._emit(OP_ADD_I64)
        return KI

    def _compile_extract_ext(self, op, ops):
        """EXTRACT fields beyond Y/M/D/H/M/S, matching the REFERENCE's
        date_part exactly (rex/core/call.py:917-960): CENTURY/DECADE/
        MILLENNIUM = trunc(year/unit); DOW = (pandas dayofweek+1)%7, i.e.
        Sunday=0 ((days+4) mod 7 — epoch day 0 was a Thursday); DOY = days
        since Jan 1 + 1; QUARTER = (month+2)/3; MICROSECOND = sub-second
        microseconds; MILLISECOND = 1000*microsecond (the reference's own
        convention, kept for parity)."""
        t0 = getattr(ops[0], "getType", lambda: None)()
        is_ts = t0 is not None and t0.getSqlType() == "TIMESTAMP"
        x = ops[0]
        field = op[len("EXTRACT_"):]

        def emit_y():
            self._emit_days(x, is_ts)
            self._emit(OP_YEAR)

        if field in ("CENTURY", "DECADE", "MILLENNIUM"):
            emit_y()
            self._emit(OP_LIT_I64, 0,
                       {"CENTURY": 100, "DECADE": 10,
                        "MILLENNIUM": 1000}[field])
            self._emit(OP_DIV_I64)
            return KI
        if field == "QUARTER":
            self._emit_days(x, is_ts)
            self._emit(OP_MONTH)
            self._emit(OP_LIT_I64, 0, 2)
            self._emit(OP_ADD_I64)
            self._emit(OP_LIT_I64, 0, 3)
            self._emit(OP_DIV_I64)
            return KI
        if field == "DOW":
            self._emit_days(x, is_ts)
            self._emit(OP_LIT_I64, 0, 4)
            self._emit(OP_ADD_I64)
            self._emit(OP_LIT_I64, 0, 7)
            self._emit(OP_FLOORMOD_I64)
            return KI
        if field == "DOY":
            # days - jan1_days + 1 (same leap-count identity as
            # FLOOR_TO_YEAR)
            self._emit_days(x, is_ts)
            emit_y()
            self._emit(OP_LIT_I64, 0, 365)
            self._emit(OP_MUL_I64)

            def leap(div):
                emit_y()
                self._emit(OP_LIT_I64, 0, 1)
                self._emit(OP_SUB_I64)
                self._emit(OP_LIT_I64, 0, div)
                self._emit(OP_DIV_I64)

            leap(4)
            self._emit(OP_ADD_I64)
            leap(100)
            self._emit(OP_SUB_I64)
            leap(400)
            self._emit(OP_ADD_I64)
            self._emit(OP_LIT_I64, 0, 719_527)
            self._emit(OP_SUB_I64)
            self._emit(OP_SUB_I64)
            self._emit(OP_LIT_I64, 0, 1)
            self._emit(OP_ADD_I64)
            return KI
        if field in ("MICROSECOND", "MILLISECOND"):
            if not is_ts:
                # a DATE has no sub-second part; 0*x keeps x's validity
                k = self.compile(x)
                if k != KI:
                    raise RexCompileError(f"{op} needs DATE/TIMESTAMP")
                self._emit(OP_LIT_I64, 0, 0)
                self._emit(OP_MUL_I64)
                return KI
            self.compile(x)
            self._emit(OP_LIT_I64, 0, 1_000_000_000)
            self._emit(OP_FLOORMOD_I64)
            self._emit(OP_LIT_I64, 0, 1000)
            self._emit(OP_DIV_I64)
            if field == "MILLISECOND":
                self._emit(OP_LIT_I64, 0, 1000)
                self._emit(OP_MUL_I64)
            return KI
        raise RexCompileError(f"operator {op} not supported on GPU path")

    def _compile_like(self, ops, mode="LIKE"):
        """LIKE/ILIKE/SIMILAR on a dict-encoded column: the SQL pattern
        (%/_ wildcards, reference rex/core/call.py SargPythonImplementation
        / re-based LIKE lowering) is matched against the (small,
        host-resident) dictionary once at compile time; the kernel-side
        predicate is an OR-chain of integer code equalities, so NULL → NULL
        falls out of EQ validity. A third operand is the ESCAPE char."""
        col, pat = ops[0], ops[1]
        esc = None
        if len(ops) > 2 and isinstance(ops[2], Literal):
            esc = ops[2].getValue()
        fn = dict_string_fn(col, self.dicts)
        if fn is None or not (isinstance(pat, Literal)
                              and isinstance(pat.getValue(), str)):
            raise RexCompileError(
                "LIKE needs <dict string expr> LIKE '<pattern>'")
        ci, f = fn
        d = self.dicts[ci]
        rx = _like_regex(pat.getValue(), esc, mode)
        matched = [i for i, s in enumerate(d)
                   if s is not None and rx.fullmatch(f(s))]
        return self._emit_code_in(ci, matched)

    def _emit_code_in(self, ci, codes):
        """predicate: column's dict code ∈ codes (OR-chain of EQ; NULL→NULL
        via EQ validity). Empty set → always-FALSE via a never-present
        code."""
        if len(codes) > 24:
            # 24 codes = 95 VM instructions, leaving headroom in the
            # 120-slot program budget (DSX_MAX_PROG) for the surrounding
            # predicate; larger dictionaries need a LUT-gather predicate
            # column (round-3)
            raise RexCompileError(
                f"predicate matches {len(codes)} dictionary entries "
                "(> VM program budget)")
        if not codes:
            self._emit(OP_COL, ci)
            self._emit(OP_LIT_I64, 0, -2)
            self._emit(OP_EQ_I64)
            return KB
        for j, code in enumerate(codes):
            self._emit(OP_COL, ci)
            self._emit(OP_LIT_I64, 0, code)
            self._emit(OP_EQ_I64)
            if j:
                self._emit(OP_OR)
        return KB

    def _compile_case(self, ops):
        # rightmost-else first; build nested SELECTs. Postfix SELECT pops
        # (cond, a, b). Emit conds/vals in order with SELECT folds from the
        # back: CASE c1 v1 c2 v2 e == SELECT(c1, v1, SELECT(c2, v2, e)).
        # Branch kinds are UNIFIED to float if any branch is float —
        # SELECT moves raw slots, so mixed int/float branches would
        # otherwise reinterpret bits downstream (same pre-scan COALESCE
        # does).
        *pairs, els = ops
        assert len(pairs) % 2 == 0
        vals = [pairs[i + 1] for i in range(0, len(pairs), 2)] + [els]
        target = KF if any(self._peek_kind(v) == KF for v in vals) else None

        def emit_chain(i):
            if i >= len(pairs):
                k = self.compile(els)
                if target == KF:
                    k = self._to_f(k)
                return k
            self.compile(pairs[i])        # cond
            kv = self.compile(pairs[i + 1])  # val
            if target == KF:
                kv = self._to_f(kv)
            emit_chain(i + 2)
            self._emit(OP_SELECT)
            return kv

        return emit_chain(0)

    def _compile_cmp(self, op, ops) -> str:
        a, b = ops
        # dict-encoded string compare: string expr chain vs string literal
        lit, sexpr = None, None
        if isinstance(b, Literal) and isinstance(b.getValue(), str):
            sexpr, lit = a, b
        elif isinstance(a, Literal) and isinstance(a.getValue(), str):
            sexpr, lit = b, a
            op = {"<": ">", ">": "<", "<=": ">=", ">=": "<="}.get(op, op)
        if sexpr is not None:
            fn = dict_string_fn(sexpr, self.dicts)
            if fn is None:
                raise RexCompileError("string compare on non-dict column")
            if op not in ("=", "<>"):
                raise RexCompileError("only =/<> on dict-encoded strings")
            ci, f = fn
            d = self.dicts[ci]
            if isinstance(sexpr, InputRef):
                # plain column: single-code compare keeps <> 3VL cheap
                code = d.index(lit.getValue()) if lit.getValue() in d else -2
                self._emit(OP_COL, ci)
                self._emit(OP_LIT_I64, 0, code)
                self._emit(_CMP[op][0])
                return KB
            matched = [c for c, s in enumerate(d)
                       if s is not None and f(s) == lit.getValue()]
            self._emit_code_in(ci, matched)
            if op == "<>":
                self._emit(OP_NOT)
            return KB
        ka = self._peek_kind(a)
        kb = self._peek_kind(b)
        if KF in (ka, kb):
            k = self.compile(a)
            self._to_f(k)
            k = self.compile(b)
            self._to_f(k)
            self._emit(_CMP[op][1])
        else:
            self.compile(a)
            self.compile(b)
            self._emit(_CMP[op][0])
        return KB

    def _peek_kind(self, expr) -> str:
        """Result kind without emitting (cheap recursive type-check)."""
        if isinstance(expr, InputRef):
            return _DSX_KIND[self.cols[expr.getIndex()].dtype]
        if isinstance(expr, Literal):
            v = expr.getValue()
            if isinstance(v, bool):
                return KB
            if isinstance(v, float):
                return KF
            return KI
        if isinstance(expr, Call):
            op = expr.getOperatorName()
            if op in _CMP or op in ("AND", "OR", "NOT", "IS NULL",
                                    "IS NOT NULL", "LIKE", "ILIKE",
                                    "SIMILAR"):
                return KB
            if op in _ARITH:
                ka = self._peek_kind(expr.getOperands()[0])
                kb = self._peek_kind(expr.getOperands()[1])
                return KF if KF in (ka, kb) else KI
            if op == "CAST":
                return _SQL_TO_KIND.get(expr.getType().getSqlType(), KF)
            if op in ("NEG", "ABS"):
                return self._peek_kind(expr.getOperands()[0])
            if op == "CASE":
                # must match _compile_case's target rule: float if ANY
                # branch value (or the else) is float
                ops_ = expr.getOperands()
                vals = [ops_[i + 1] for i in range(0, len(ops_) - 1, 2)]
                if len(ops_) % 2 == 1:
                    vals.append(ops_[-1])
                if any(self._peek_kind(v) == KF for v in vals):
                    return KF
                return self._peek_kind(ops_[1])
            if op == "MOD":
                ka = self._peek_kind(expr.getOperands()[0])
                kb = self._peek_kind(expr.getOperands()[1])
                return KF if KF in (ka, kb) else KI
            if op in ("EXTRACT_YEAR", "EXTRACT_MONTH", "EXTRACT_DAY",
                      "YEAR", "MONTH", "DAY", "DAYOFMONTH") \
                    or op.startswith("EXTRACT_") \
                    or op.startswith("FLOOR_TO_") \
                    or op.startswith("CEIL_TO_") or op == "LAST_DAY":
                return KI
            if op == "COALESCE":
                kids = [self._peek_kind(o) for o in expr.getOperands()]
                return KF if KF in kids else kids[-1]
            if op == "NULLIF":
                return self._peek_kind(expr.getOperands()[0])
        return KF


def compile_expr(expr, cols, dictionaries=None):
    """Returns (prog_tuple (for Runtime.make_prog), result_kind). Cached on
    the expression object — programs depend only on the column dtypes and
    dictionaries, which are stable across the steps of a cached plan."""
    key = (tuple(c.dtype for c in cols),
           tuple(id(d) for d in (dictionaries or [])))
    hit = getattr(expr, "_dsx_compiled", None)
    if hit is not None and hit[0] == key:
        return hit[1], hit[2]
    c = RexCompiler(cols, dictionaries)
    kind = c.compile(expr)
    try:
        expr._dsx_compiled = (key, c.prog, kind)
    except Exception:
        pass
    return c.prog, kind


def scalar_literal(expr):
    """If expr is a scalar boolean literal, return its Python value
    (filter_or_scalar short-circuit, reference filter.py:31-36)."""
    if isinstance(expr, Literal):
        v = expr.getValue()
        if isinstance(v, bool) or v is None:
            return bool(v) if v is not None else False
        if isinstance(v, (int, float)):
            return bool(v)
    return None
