"""CPU verification of the FLOOR/CEIL(x TO unit) / EXTRACT(DATE) VM
lowerings (rex.py _compile_dt_trunc): the emitted stack programs are
interpreted here with the same integer semantics the device VM uses
(csrc/dsxhip.hip BIN_F cases) and compared against pandas calendar truth —
no GPU needed, the opcodes themselves are already GPU-parity-tested.
Reference behavior: dask_sql rex/core/call.py CeilFloorDatetimeOperation /
ExtractOperation."""
import types

import numpy as np
import pandas as pd
import pytest

from dask_sql_amd import runtime as rt
from dask_sql_amd.physical import rex as R
from dask_sql_amd.planner.plan import Call, InputRef, Literal, SqlType

DAY_NS = 86_400_000_000_000


def _civil_from_days(days):
    d = np.datetime64(int(days), "D").astype(object)
    return d.year, d.month, d.day


def interp_prog(prog, x):
    """Interpret a rex VM program over one int64 value (the lowerings under
    test only use the integer subset; semantics mirror dsxhip.hip)."""
    def tdiv(a, b):
        q = abs(a) // abs(b)
        return q if (a >= 0) == (b >= 0) else -q

    st = []
    for op, arg0, imm in prog:
        if op == R.OP_COL:
            st.append(x)
        elif op == R.OP_LIT_I64:
            st.append(imm)
        elif op == R.OP_ADD_I64:
            b, a = st.pop(), st.pop()
            st.append(a + b)
        elif op == R.OP_SUB_I64:
            b, a = st.pop(), st.pop()
            st.append(a - b)
        elif op == R.OP_MUL_I64:
            b, a = st.pop(), st.pop()
            st.append(a * b)
        elif op == R.OP_DIV_I64:
            b, a = st.pop(), st.pop()
            st.append(tdiv(a, b) if b else 0)
        elif op == R.OP_MOD_I64:
            b, a = st.pop(), st.pop()
            st.append(a - tdiv(a, b) * b if b else 0)
        elif op == R.OP_FLOORMOD_I64:
            b, a = st.pop(), st.pop()
            st.append(((a - tdiv(a, b) * b) + b) % b if b else 0)
        elif op == R.OP_DAY:
            st.append(_civil_from_days(st.pop())[2])
        elif op == R.OP_MONTH:
            st.append(_civil_from_days(st.pop())[1])
        elif op == R.OP_YEAR:
            st.append(_civil_from_days(st.pop())[0])
        else:
            raise AssertionError(f"unexpected opcode {op}")
    assert len(st) == 1
    return st[0]


def _compile(op_name, sql_type):
    col = types.SimpleNamespace(dtype=rt.I64)
    e = Call(op_name, [InputRef(0, SqlType(sql_type))], SqlType(sql_type))
    c = R.RexCompiler([col])
    k = c.compile(e)
    assert k == R.KI
    return c.prog


TS_SAMPLES = [
    "2021-02-01 13:45:12.345",
    "2020-02-29 23:59:59.999",   # leap day
    "2000-12-31 00:00:00",
    "1970-01-01 00:00:00",
    "1969-07-20 20:17:40",       # pre-epoch
    "1900-03-01 04:00:00",       # 1900 not a leap year
    "2200-03-01 12:00:00",       # past the next two century rules
    "2021-01-01 00:00:00",       # already on every boundary
]


@pytest.mark.parametrize("unit,pd_how", [
    ("FLOOR_TO_DAY", "D"), ("FLOOR_TO_HOUR", "h"),
    ("FLOOR_TO_MINUTE", "min"), ("FLOOR_TO_SECOND", "s"),
])
def test_floor_subday_vs_pandas(unit, pd_how):
    prog = _compile(unit, "TIMESTAMP")
    for s in TS_SAMPLES:
        ts = pd.Timestamp(s)
        got = interp_prog(prog, ts.value)
        assert got == ts.floor(pd_how).value, (unit, s)


@pytest.mark.parametrize("unit,pd_how", [
    ("CEIL_TO_DAY", "D"), ("CEIL_TO_HOUR", "h"),
    ("CEIL_TO_MINUTE", "min"), ("CEIL_TO_SECOND", "s"),
])
def test_ceil_subday_vs_pandas(unit, pd_how):
    prog = _compile(unit, "TIMESTAMP")
    for s in TS_SAMPLES:
        ts = pd.Timestamp(s)
        got = interp_prog(prog, ts.value)
        assert got == ts.ceil(pd_how).value, (unit, s)


def test_floor_month_year_vs_pandas():
    pm = _compile("FLOOR_TO_MONTH", "TIMESTAMP")
    py = _compile("FLOOR_TO_YEAR", "TIMESTAMP")
    for s in TS_SAMPLES:
        ts = pd.Timestamp(s)
        want_m = pd.Timestamp(year=ts.year, month=ts.month, day=1).value
        want_y = pd.Timestamp(year=ts.year, month=1, day=1).value
        assert interp_prog(pm, ts.value) == want_m, ("MONTH", s)
        assert interp_prog(py, ts.value) == want_y, ("YEAR", s)


def test_floor_on_date_column():
    # DATE input is day-count; DAY floor is the identity, MONTH/YEAR floors
    # return the first-of-period day count
    pd_day = _compile("FLOOR_TO_DAY", "DATE")
    pm = _compile("FLOOR_TO_MONTH", "DATE")
    py = _compile("FLOOR_TO_YEAR", "DATE")
    for s in ("2021-02-17", "1969-12-31", "2000-02-29"):
        days = (pd.Timestamp(s) - pd.Timestamp(0)).days
        assert interp_prog(pd_day, days) == days
        ts = pd.Timestamp(s)
        want_m = (pd.Timestamp(year=ts.year, month=ts.month, day=1)
                  - pd.Timestamp(0)).days
        want_y = (pd.Timestamp(year=ts.year, month=1, day=1)
                  - pd.Timestamp(0)).days
        assert interp_prog(pm, days) == want_m
        assert interp_prog(py, days) == want_y


def test_extract_date_vs_pandas():
    prog = _compile("EXTRACT_DATE", "TIMESTAMP")
    for s in TS_SAMPLES:
        ts = pd.Timestamp(s)
        want = (ts.normalize() - pd.Timestamp(0)).days
        assert interp_prog(prog, ts.value) == want, s


def test_ceil_month_year_vs_pandas():
    """Calendar CEIL = next-period start of (x - 1 tick): boundary inputs
    are fixed points, everything else rounds up (full interpreter — the
    lowering uses EQ-free decrement arithmetic)."""
    from tests.vm_interp import interp

    def run(prog, v):
        got, ok = interp(prog, [(np.array([v], dtype=np.int64), None)], 0)
        assert ok
        return got

    pm = _compile("CEIL_TO_MONTH", "TIMESTAMP")
    py = _compile("CEIL_TO_YEAR", "TIMESTAMP")
    for s in TS_SAMPLES:
        ts = pd.Timestamp(s)
        wm = ts if ts == ts.to_period("M").start_time \
            else (ts.to_period("M") + 1).start_time
        wy = ts if ts == ts.to_period("Y").start_time \
            else (ts.to_period("Y") + 1).start_time
        assert run(pm, ts.value) == wm.value, ("M", s)
        assert run(py, ts.value) == wy.value, ("Y", s)
    pmd = _compile("CEIL_TO_MONTH", "DATE")
    pyd = _compile("CEIL_TO_YEAR", "DATE")
    for s in ("2021-02-17", "2021-02-01", "2020-12-31", "1969-07-20",
              "2000-02-29"):
        ts = pd.Timestamp(s)
        days = (ts - pd.Timestamp(0)).days
        wm = ts if ts == ts.to_period("M").start_time \
            else (ts.to_period("M") + 1).start_time
        wy = ts if ts == ts.to_period("Y").start_time \
            else (ts.to_period("Y") + 1).start_time
        assert run(pmd, days) == (wm - pd.Timestamp(0)).days, ("Md", s)
        assert run(pyd, days) == (wy - pd.Timestamp(0)).days, ("Yd", s)


def test_extract_extended_vs_reference_semantics():
    """CENTURY/DECADE/MILLENNIUM/DOW/DOY/QUARTER/MICROSECOND/MILLISECOND
    against the reference's date_part definitions (rex/core/call.py:917):
    CENTURY = trunc(year/100), DOW = (pandas dayofweek+1)%7,
    MILLISECOND = 1000*microsecond (reference convention)."""
    for s in TS_SAMPLES:
        ts = pd.Timestamp(s)
        cases = {
            "EXTRACT_CENTURY": int(ts.year / 100),
            "EXTRACT_DECADE": int(ts.year / 10),
            "EXTRACT_MILLENNIUM": int(ts.year / 1000),
            "EXTRACT_DOW": (ts.dayofweek + 1) % 7,
            "EXTRACT_DOY": ts.dayofyear,
            "EXTRACT_QUARTER": ts.quarter,
            "EXTRACT_MICROSECOND": ts.microsecond,
            "EXTRACT_MILLISECOND": 1000 * ts.microsecond,
        }
        for op_name, want in cases.items():
            prog = _compile(op_name, "TIMESTAMP")
            assert interp_prog(prog, ts.value) == want, (op_name, s)


def test_extract_extended_on_date():
    for s in ("2021-02-17", "2000-02-29", "1971-01-01"):
        ts = pd.Timestamp(s)
        days = (ts - pd.Timestamp(0)).days
        for op_name, want in {
            "EXTRACT_DOW": (ts.dayofweek + 1) % 7,
            "EXTRACT_DOY": ts.dayofyear,
            "EXTRACT_QUARTER": ts.quarter,
            "EXTRACT_CENTURY": int(ts.year / 100),
            "EXTRACT_MICROSECOND": 0,
            "EXTRACT_MILLISECOND": 0,
        }.items():
            prog = _compile(op_name, "DATE")
            assert interp_prog(prog, days) == want, (op_name, s)


def test_last_day_vs_pandas():
    """LAST_DAY = x + MonthEnd(1), incl. the anchor ROLL when x is already
    a month end (reference call.py last_day)."""
    from pandas.tseries.offsets import MonthEnd

    from tests.vm_interp import interp

    def run(prog, v):
        got, ok = interp(prog, [(np.array([v], dtype=np.int64), None)], 0)
        assert ok
        return got

    pt = _compile("LAST_DAY", "TIMESTAMP")
    for s in TS_SAMPLES:
        ts = pd.Timestamp(s)
        want = ts + MonthEnd(1)
        assert run(pt, ts.value) == want.value, s
    pdt = _compile("LAST_DAY", "DATE")
    for s in ("2021-01-15", "2021-01-31", "2020-02-29", "1969-12-15"):
        ts = pd.Timestamp(s)
        days = (ts - pd.Timestamp(0)).days
        want = (ts + MonthEnd(1) - pd.Timestamp(0)).days
        assert run(pdt, days) == want, s


def test_reference_unit_golden_vectors():
    """The reference's own unit expectations (tests/unit/test_call.py:202-
    238, datetime(2021,10,3,15,53,42,47)) — EXTRACT fields and sub-day
    FLOOR/CEIL including MILLISECOND."""
    import datetime as dtm

    from tests.vm_interp import interp

    def run(op_name, ty, v):
        prog = _compile(op_name, ty)
        got, ok = interp(prog, [(np.array([v], dtype=np.int64), None)], 0)
        assert ok
        return got

    ts = pd.Timestamp(dtm.datetime(2021, 10, 3, 15, 53, 42, 47))
    v = ts.value
    assert run("EXTRACT_CENTURY", "TIMESTAMP", v) == 20
    assert run("EXTRACT_DECADE", "TIMESTAMP", v) == 202
    assert run("EXTRACT_DOW", "TIMESTAMP", v) == 0
    assert run("EXTRACT_DOY", "TIMESTAMP", v) == 276
    assert run("EXTRACT_MILLENNIUM", "TIMESTAMP", v) == 2
    assert run("EXTRACT_MICROSECOND", "TIMESTAMP", v) == 47
    assert run("EXTRACT_MILLISECOND", "TIMESTAMP", v) == 47000
    assert run("EXTRACT_QUARTER", "TIMESTAMP", v) == 4
    d = dtm.datetime
    assert run("CEIL_TO_DAY", "TIMESTAMP", v) == \
        pd.Timestamp(d(2021, 10, 4)).value
    assert run("CEIL_TO_HOUR", "TIMESTAMP", v) == \
        pd.Timestamp(d(2021, 10, 3, 16)).value
    assert run("CEIL_TO_MINUTE", "TIMESTAMP", v) == \
        pd.Timestamp(d(2021, 10, 3, 15, 54)).value
    assert run("CEIL_TO_SECOND", "TIMESTAMP", v) == \
        pd.Timestamp(d(2021, 10, 3, 15, 53, 43)).value
    assert run("CEIL_TO_MILLISECOND", "TIMESTAMP", v) == \
        pd.Timestamp(d(2021, 10, 3, 15, 53, 42, 1000)).value
    assert run("FLOOR_TO_DAY", "TIMESTAMP", v) == \
        pd.Timestamp(d(2021, 10, 3)).value
    assert run("FLOOR_TO_MILLISECOND", "TIMESTAMP", v) == \
        pd.Timestamp(d(2021, 10, 3, 15, 53, 42)).value
