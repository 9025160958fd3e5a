"""Synthetic input generators for the benchmark configs.

All inputs numpy.random.default_rng(seed) with seed 42 by default; sizes
scalable for tests. Shared by bench.py, tests/ and the oracle checks —
harness code, not product code.
"""
from __future__ import annotations

import numpy as np
import pandas as pd

SEED = 42

# dictionary code maps for the synthetic TPC-H columns
RETURNFLAG_DICT = ["A", "N", "R"]
LINESTATUS_DICT = ["F", "O"]
MKTSEGMENT_DICT = ["BUILDING", "AUTOMOBILE", "FURNITURE", "HOUSEHOLD", "MACHINERY"]

D_EPOCH = np.datetime64("1970-01-01")


def _days(s: str) -> int:
    return int((np.datetime64(s) - D_EPOCH).astype(int))


def gen_c1(n=1_000_000, n_groups=1_000, seed=SEED):
    """C1: 1M-row 2-column frame: key in [0,1000), x uniform [0,1)."""
    rng = np.random.default_rng(seed)
    key = rng.integers(0, n_groups, size=n, dtype=np.int64)
    x = rng.random(size=n)
    return key, x


def gen_c2(n=100_000_000, n_groups=1_000_000, seed=SEED):
    """C2: 100M rows int64 key ~ U[0, n_groups), fp64 val ~ U[0,1)."""
    rng = np.random.default_rng(seed)
    key = rng.integers(0, n_groups, size=n, dtype=np.int64)
    val = rng.random(size=n)
    return key, val


def gen_c3(n_build=10_000_000, n_probe=100_000_000, seed=SEED):
    """C3: build = shuffled arange (unique), probe = choice of build keys."""
    rng = np.random.default_rng(seed)
    build_key = rng.permutation(n_build).astype(np.int64)
    probe_key = rng.integers(0, n_build, size=n_probe, dtype=np.int64)  # 100% hit
    build_val = rng.random(size=n_build)
    probe_val = rng.random(size=n_probe)
    return build_key, build_val, probe_key, probe_val


def gen_lineitem_q1(n=59_986_052, seed=SEED) -> pd.DataFrame:
    """C4: synthetic TPC-H SF10 lineitem for Q1 (SURVEY §8d distributions).

    Columns: l_quantity f64 (uniform int 1-50), l_extendedprice f64 (derived),
    l_discount f64 (uniform 0-0.1), l_tax f64 (0-0.08), l_returnflag i8 code,
    l_linestatus i8 code, l_shipdate date32 (uniform 1992-1998).
    """
    rng = np.random.default_rng(seed)
    qty = rng.integers(1, 51, size=n).astype(np.float64)
    extprice = qty * (90000.0 + 100000.0 * rng.random(size=n)) / 50.0
    discount = np.round(rng.random(size=n) * 0.10, 2)
    tax = np.round(rng.random(size=n) * 0.08, 2)
    returnflag = rng.integers(0, len(RETURNFLAG_DICT), size=n).astype(np.int8)
    linestatus = rng.integers(0, len(LINESTATUS_DICT), size=n).astype(np.int8)
    shipdate = rng.integers(
        _days("1992-01-01"), _days("1998-12-01"), size=n
    ).astype(np.int32)
    return pd.DataFrame(
        {
            "l_quantity": qty,
            "l_extendedprice": extprice,
            "l_discount": discount,
            "l_tax": tax,
            "l_returnflag": returnflag,
            "l_linestatus": linestatus,
            "l_shipdate": shipdate,
        }
    )


def gen_q3(sf_rows=(1_500_000, 15_000_000, 60_000_000), seed=SEED):
    """C5: synthetic TPC-H SF10 customer / orders / lineitem for Q3.

    customer: c_custkey i64 (unique), c_mktsegment i8 code (uniform of 5;
      code 0 = BUILDING → 1/5 selectivity).
    orders: o_orderkey i64 (unique), o_custkey i64 ~ U[customers],
      o_orderdate date32 uniform 1992-1998 (≈half < 1995-03-15),
      o_shippriority i32 = 0.
    lineitem: l_orderkey i64 ~ U[orders], l_extendedprice / l_discount f64,
      l_shipdate date32 uniform 1992-1998 (≈half > 1995-03-15).
    """
    n_cust, n_ord, n_li = sf_rows
    rng = np.random.default_rng(seed)
    cust = pd.DataFrame(
        {
            "c_custkey": np.arange(n_cust, dtype=np.int64),
            "c_mktsegment": rng.integers(0, 5, size=n_cust).astype(np.int8),
        }
    )
    orders = pd.DataFrame(
        {
            "o_orderkey": np.arange(n_ord, dtype=np.int64),
            "o_custkey": rng.integers(0, n_cust, size=n_ord, dtype=np.int64),
            "o_orderdate": rng.integers(
                _days("1992-01-01"), _days("1998-12-01"), size=n_ord
            ).astype(np.int32),
            "o_shippriority": np.zeros(n_ord, dtype=np.int32),
        }
    )
    li = pd.DataFrame(
        {
            "l_orderkey": rng.integers(0, n_ord, size=n_li, dtype=np.int64),
            "l_extendedprice": 900.0 + 100000.0 * rng.random(size=n_li),
            "l_discount": np.round(rng.random(size=n_li) * 0.10, 2),
            "l_shipdate": rng.integers(
                _days("1992-01-01"), _days("1998-12-01"), size=n_li
            ).astype(np.int32),
        }
    )
    return cust, orders, li


# The REAL TPC-H Q3 text (the timed query must be the
# reference's own literals — 'BUILDING' and DATE '1995-03-15' — not
# pre-encoded codes/day-ints; literal encoding is the planner's job).
Q3_SQL = """SELECT l_orderkey, SUM(l_extendedprice*(1-l_discount)) AS revenue,
 o_orderdate, o_shippriority
 FROM customer, orders, lineitem
 WHERE c_mktsegment = 'BUILDING' AND c_custkey = o_custkey
 AND l_orderkey = o_orderkey AND o_orderdate < DATE '1995-03-15'
 AND l_shipdate > DATE '1995-03-15'
 GROUP BY l_orderkey, o_orderdate, o_shippriority
 ORDER BY revenue DESC, o_orderdate LIMIT 10"""

# Q1 with the reference literals (DATE '1998-12-01' - INTERVAL '90' DAY
# folds to 1998-09-02 at plan time, like DataFusion's SimplifyExpressions)
Q1_SQL = """SELECT l_returnflag, l_linestatus, SUM(l_quantity) AS sum_qty,
 SUM(l_extendedprice) AS sum_base_price,
 SUM(l_extendedprice*(1-l_discount)) AS sum_disc_price,
 SUM(l_extendedprice*(1-l_discount)*(1+l_tax)) AS sum_charge,
 AVG(l_quantity) AS avg_qty, AVG(l_extendedprice) AS avg_price,
 AVG(l_discount) AS avg_disc, COUNT(*) AS count_order
 FROM lineitem WHERE l_shipdate <= DATE '1998-12-01' - INTERVAL '90' DAY
 GROUP BY l_returnflag, l_linestatus"""


def register_q3_tables(ctx, cust, orders, li, persist=False):
    """Register the Q3 tables with their REAL types: c_mktsegment as a
    dictionary-encoded string column, o_orderdate/l_shipdate as DATE — so
    Q3_SQL's 'BUILDING' / DATE literals plan exactly as on the reference."""
    ctx.create_table("customer", cust, persist=persist,
                     dictionaries={"c_mktsegment": MKTSEGMENT_DICT})
    ctx.create_table("orders", orders, persist=persist,
                     date_columns=("o_orderdate",))
    ctx.create_table("lineitem", li, persist=persist,
                     date_columns=("l_shipdate",))


def register_q1_table(ctx, li, persist=False):
    """Register the Q1 lineitem with real types: returnflag/linestatus as
    dictionary strings, l_shipdate as DATE."""
    ctx.create_table("lineitem", li, persist=persist,
                     date_columns=("l_shipdate",),
                     dictionaries={"l_returnflag": RETURNFLAG_DICT,
                                   "l_linestatus": LINESTATUS_DICT})
