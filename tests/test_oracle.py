"""Pin the oracle against the reference's golden vectors.

Every case cites the reference integration test whose literal expected frame
it restates (tests/golden/golden_cases.json)."""
import numpy as np
import pandas as pd
import pytest

from oracle.frame import oracle_filter, oracle_groupby, oracle_join
from oracle.tpch import oracle_c1_c2_groupby, oracle_c3_join, oracle_q1, oracle_q3
from tests.conftest import assert_frame_close, golden_expected


def test_groupby_simple(user_table_1):
    # test_groupby.py:25-36
    out = oracle_groupby(user_table_1, ["user_id"], [("b", "S", "sum", None, False)])
    out = out.sort_values("user_id").reset_index(drop=True)
    assert_frame_close(out, golden_expected("groupby_simple"))


def test_groupby_all(user_table_1):
    # test_groupby.py:70-78 — SUM(2) = sum of literal-2 column
    df = user_table_1.assign(__lit2__=2)
    out = oracle_groupby(
        df, [], [("b", "S", "sum", None, False), ("__lit2__", "X", "sum", None, False)]
    )
    assert_frame_close(out, golden_expected("groupby_all"))


def test_groupby_filtered_full(user_table_1):
    # test_groupby.py:108-118
    df = user_table_1.assign(__f__=user_table_1["user_id"] == 2)
    out = oracle_groupby(
        df, [], [("b", "S1", "sum", "__f__", False), ("b", "S2", "sum", None, False)]
    )
    out = out[["S1", "S2"]]
    assert_frame_close(out, golden_expected("groupby_filtered_full"))


def test_groupby_filtered_grouped(user_table_1):
    # test_groupby.py:121-140 — filtered agg leaves NULL for groups with no
    # qualifying rows; the non-filtered pass keeps all groups.
    df = user_table_1.assign(__f__=user_table_1["user_id"] == 2)
    out = oracle_groupby(
        df,
        ["user_id"],
        [("b", "S1", "sum", "__f__", False), ("b", "S2", "sum", None, False)],
    )
    out = out.sort_values("user_id").reset_index(drop=True)
    assert_frame_close(out, golden_expected("groupby_filtered_grouped"))


def _named(df, names):
    df = df.copy()
    df.columns = names
    return df


def test_join_inner(user_table_1, user_table_2):
    # test_join.py:14-43; output = project [lhs.user_id, lhs.b, rhs.c]
    out = oracle_join(user_table_1, user_table_2, [0], [0], "INNER")
    out = _named(out[["lhs_0", "lhs_1", "rhs_1"]], ["user_id", "b", "c"])
    assert_frame_close(
        out, golden_expected("join_inner"), sort_by=["user_id", "b", "c"]
    )


def test_join_outer(user_table_1, user_table_2):
    # test_join.py:46-66 — FULL OUTER fills NaN (not NA)
    out = oracle_join(user_table_1, user_table_2, [0], [0], "FULL")
    out = _named(out[["lhs_0", "lhs_1", "rhs_1"]], ["user_id", "b", "c"])
    assert_frame_close(
        out, golden_expected("join_outer"), sort_by=["user_id", "b", "c"]
    )


def test_join_left(user_table_1, user_table_2):
    # test_join.py:69-88
    out = oracle_join(user_table_1, user_table_2, [0], [0], "LEFT")
    out = _named(out[["lhs_0", "lhs_1", "rhs_1"]], ["user_id", "b", "c"])
    assert_frame_close(
        out, golden_expected("join_left"), sort_by=["user_id", "b", "c"]
    )


def test_join_residual(user_table_1, user_table_2):
    # test_join.py:210-225 — equi key + residual rhs.c - lhs.b >= 0
    out = oracle_join(
        user_table_1,
        user_table_2,
        [0],
        [0],
        "INNER",
        residual=lambda d: (d["rhs_1"] - d["lhs_1"]) >= 0,
    )
    out = _named(out, ["lhs.user_id", "b", "rhs.user_id", "c"])
    assert_frame_close(
        out, golden_expected("join_residual"), sort_by=["lhs.user_id", "b", "c"]
    )


def test_join_conditional(df_simple):
    # test_join.py:190-207 — no equi keys: cross join + residual filter
    out = oracle_join(
        df_simple,
        df_simple,
        [],
        [],
        "INNER",
        residual=lambda d: (d["lhs_0"] < d["rhs_1"]) & (d["lhs_1"] < d["rhs_0"]),
    )
    out = _named(out, ["lhs.a", "lhs.b", "rhs.a", "rhs.b"])
    assert_frame_close(
        out, golden_expected("join_conditional"), sort_by=["lhs.a", "rhs.a"]
    )


def test_join_literal_true(user_table_1, user_table_2):
    # test_join.py:227-246 — ON TRUE = cross join
    out = oracle_join(user_table_1, user_table_2, [], [], "INNER")
    out = _named(out, ["lhs.user_id", "b", "rhs.user_id", "c"])
    assert_frame_close(
        out,
        golden_expected("join_literal_true"),
        sort_by=["lhs.user_id", "b", "rhs.user_id", "c"],
    )


def test_join_literal_false(user_table_1, user_table_2):
    # test_join.py:248-259 — ON FALSE = empty result
    crossed = oracle_join(user_table_1, user_table_2, [], [], "INNER")
    out = oracle_filter(crossed, False)
    assert len(out) == 0
    assert list(out.columns) == ["lhs_0", "lhs_1", "rhs_0", "rhs_1"]


def test_filter_with_nan():
    # test_filter.py:52-59 — NULL comparison is not True → filtered out
    df = pd.DataFrame({"c": pd.array([3, pd.NA, 1], dtype="UInt8")})
    out = oracle_filter(df, df["c"] == 3)
    assert out["c"].tolist() == [3]


def test_filter_null_is_false(df700):
    # filter.py:39 — NULL → False; also plain float predicate on seed-42 df
    out = oracle_filter(df700, df700["a"] < 2)
    pd.testing.assert_frame_equal(
        out.reset_index(drop=True), df700[df700["a"] < 2].reset_index(drop=True)
    )
    df = pd.DataFrame({"x": [1.0, np.nan, 3.0]})
    cond = pd.Series([True, pd.NA, True], dtype="boolean")
    out = oracle_filter(df, cond)
    assert out["x"].tolist() == [1.0, 3.0]


def test_sum_all_null_is_null():
    # aggregate.py:486-493 custom_sum min_count=1
    df = pd.DataFrame({"k": [1, 1, 2], "v": [np.nan, np.nan, 1.0]})
    out = oracle_groupby(df, ["k"], [("v", "s", "sum", None, False)])
    out = out.sort_values("k").reset_index(drop=True)
    assert np.isnan(out["s"][0]) and out["s"][1] == 1.0


def test_groupby_null_group_kept():
    # aggregate.py:575-577 dropna=False
    df = pd.DataFrame({"k": [1.0, np.nan, np.nan], "v": [1.0, 2.0, 3.0]})
    out = oracle_groupby(df, ["k"], [("v", "s", "sum", None, False)])
    assert len(out) == 2
    s_null = out[out["k"].isna()]["s"].iloc[0]
    assert s_null == 5.0


def test_groupby_distinct():
    # aggregate.py:562-565 DISTINCT via drop_duplicates
    df = pd.DataFrame({"k": [1, 1, 1, 2], "v": [5, 5, 7, 5]})
    out = oracle_groupby(df, ["k"], [("v", "s", "sum", None, True)])
    out = out.sort_values("k").reset_index(drop=True)
    assert out["s"].tolist() == [12, 5]


# --- config-shaped smoke checks (small sizes) -----------------------------


def test_c1_small():
    from datagen import gen_c1

    key, x = gen_c1(n=10_000, n_groups=100)
    out = oracle_c1_c2_groupby(key, x, predicate=False)
    assert len(out) == 100
    assert np.isclose(out["s"].sum(), x.sum(), rtol=1e-12)
    assert out["c"].sum() == 10_000


def test_c2_small():
    from datagen import gen_c2

    key, val = gen_c2(n=50_000, n_groups=1_000)
    out = oracle_c1_c2_groupby(key, val, predicate=True)
    sel = val < 0.5
    assert out["c"].sum() == sel.sum()
    assert np.isclose(out["s"].sum(), val[sel].sum(), rtol=1e-12)


def test_c3_small():
    from datagen import gen_c3

    bk, bv, pk, pv = gen_c3(n_build=1_000, n_probe=5_000)
    out = oracle_c3_join(bk, pk, bv, pv)
    assert len(out) == 5_000  # 100% hit rate, unique build keys
    # every probe row carries its build val
    bmap = dict(zip(bk.tolist(), bv.tolist()))
    got = out.sort_values("pv").reset_index(drop=True)
    exp_bv = got["key"].map(bmap)
    assert np.allclose(got["bv"], exp_bv)


def test_q1_small():
    from datagen import gen_lineitem_q1

    li = gen_lineitem_q1(n=20_000)
    out = oracle_q1(li)
    assert 1 <= len(out) <= 6  # 3 returnflags × 2 linestatus
    assert out["count_order"].sum() == (li["l_shipdate"] <= 10471).sum()


def test_q3_small():
    from datagen import gen_q3

    cust, orders, li = gen_q3(sf_rows=(1_000, 5_000, 20_000))
    out = oracle_q3(cust, orders, li)
    assert len(out) <= 10
    assert list(out.columns) == ["l_orderkey", "revenue", "o_orderdate", "o_shippriority"]
    # revenue sorted descending
    assert (np.diff(out["revenue"].to_numpy()) <= 1e-9).all()


def test_oracle_global_agg_empty_selection():
    """SQL global aggregate over zero rows -> one row: COUNT 0, SUM/AVG NULL
    (reference aggregate.py:251 whole-frame agg; test_groupby.py empty-filter
    cases)."""
    df = pd.DataFrame({"v": [1.0, 2.0], "f": [False, False]})
    out = oracle_groupby(df[df.f], [], [
        ("v", "c", "count", None, False),
        ("v", "s", "sum", None, False),
        ("v", "a", "avg", None, False),
    ])
    assert len(out) == 1
    assert out["c"].iloc[0] == 0
    assert np.isnan(out["s"].iloc[0]) and np.isnan(out["a"].iloc[0])
