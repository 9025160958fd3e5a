"""Oracle pipelines for the benchmark configs (BASELINE.json configs[0..4]).

Each pipeline composes the restated operator semantics in oracle/frame.py the
same way the reference's RelConverter recursion would (SURVEY.md §3 call
stacks). TEST INFRASTRUCTURE / CPU BASELINE ONLY — oracle/__init__.py header.
"""
from __future__ import annotations

import numpy as np
import pandas as pd

from oracle.frame import oracle_filter, oracle_groupby, oracle_join

# Date constants as date32 day-ints (days since 1970-01-01), matching the
# reference's Date32 literal → day compare (rex/core/literal.py:149-151).
D_1995_03_15 = (np.datetime64("1995-03-15") - np.datetime64("1970-01-01")).astype(int)
D_1998_09_02 = (np.datetime64("1998-09-02") - np.datetime64("1970-01-01")).astype(int)


def oracle_c1_c2_groupby(key: np.ndarray, x: np.ndarray, predicate: bool = True):
    """SELECT key, SUM(x) AS s, COUNT(*) AS c FROM t [WHERE x < 0.5] GROUP BY key."""
    df = pd.DataFrame({"key": key, "x": x})
    if predicate:
        df = oracle_filter(df, df["x"] < 0.5)
    out = oracle_groupby(
        df,
        ["key"],
        [("x", "s", "sum", None, False), ("x", "c", "count", None, False)],
    )
    return out.sort_values("key").reset_index(drop=True)


def oracle_c3_join(build_key, probe_key, build_val, probe_val):
    """SELECT p.key, p.v, b.v FROM probe p JOIN build b ON p.key = b.key."""
    probe = pd.DataFrame({"key": probe_key, "pv": probe_val})
    build = pd.DataFrame({"key": build_key, "bv": build_val})
    out = oracle_join(probe, build, [0], [0], "INNER")
    out.columns = ["key", "pv", "bkey", "bv"]
    return out.drop(columns=["bkey"])


def oracle_q1(li: pd.DataFrame) -> pd.DataFrame:
    """TPC-H Q1 over the synthetic lineitem of datagen.gen_lineitem.

    returnflag/linestatus are dictionary codes (i8); shipdate is date32
    day-ints. Group keys returned as codes, sorted.
    """
    df = oracle_filter(li, li["l_shipdate"] <= D_1998_09_02)
    df = df.assign(
        disc_price=df["l_extendedprice"] * (1 - df["l_discount"]),
    )
    df = df.assign(charge=df["disc_price"] * (1 + df["l_tax"]))
    out = oracle_groupby(
        df,
        ["l_returnflag", "l_linestatus"],
        [
            ("l_quantity", "sum_qty", "sum", None, False),
            ("l_extendedprice", "sum_base_price", "sum", None, False),
            ("disc_price", "sum_disc_price", "sum", None, False),
            ("charge", "sum_charge", "sum", None, False),
            ("l_quantity", "avg_qty", "avg", None, False),
            ("l_extendedprice", "avg_price", "avg", None, False),
            ("l_discount", "avg_disc", "avg", None, False),
            ("l_quantity", "count_order", "count", None, False),
        ],
    )
    return out.sort_values(["l_returnflag", "l_linestatus"]).reset_index(drop=True)


def oracle_q3(cust: pd.DataFrame, orders: pd.DataFrame, li: pd.DataFrame) -> pd.DataFrame:
    """TPC-H Q3 (customer ⋈ orders ⋈ lineitem + filters + groupby + top-10).

    mktsegment code 0 == 'BUILDING' in datagen. Returns the full grouped
    frame sorted by (revenue desc, o_orderdate asc) LIMIT 10 — the ≤G-row
    ORDER BY runs on host per SURVEY §8f1.
    """
    c = oracle_filter(cust, cust["c_mktsegment"] == 0)[["c_custkey"]]
    o = oracle_filter(orders, orders["o_orderdate"] < D_1995_03_15)
    l = oracle_filter(li, li["l_shipdate"] > D_1995_03_15)

    co = oracle_join(
        o[["o_orderkey", "o_custkey", "o_orderdate", "o_shippriority"]],
        c,
        [1],
        [0],
        "INNER",
    )
    co.columns = ["o_orderkey", "o_custkey", "o_orderdate", "o_shippriority", "c_custkey"]
    col = oracle_join(
        l[["l_orderkey", "l_extendedprice", "l_discount"]],
        co[["o_orderkey", "o_orderdate", "o_shippriority"]],
        [0],
        [0],
        "INNER",
    )
    col.columns = [
        "l_orderkey", "l_extendedprice", "l_discount",
        "o_orderkey", "o_orderdate", "o_shippriority",
    ]
    col = col.assign(revenue=col["l_extendedprice"] * (1 - col["l_discount"]))
    out = oracle_groupby(
        col,
        ["l_orderkey", "o_orderdate", "o_shippriority"],
        [("revenue", "revenue", "sum", None, False)],
    )
    out = out[["l_orderkey", "revenue", "o_orderdate", "o_shippriority"]]
    out = out.sort_values(
        ["revenue", "o_orderdate"], ascending=[False, True], kind="mergesort"
    ).head(10)
    return out.reset_index(drop=True)
