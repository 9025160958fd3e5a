"""Mini TPC-H suite (-m gpu): real query shapes beyond Q1/Q3 — filter-agg
(Q6), EXISTS (Q4), join + CASE aggregation (Q12), CASE ratio (Q14) — on
small synthetic tables, checked against pandas restatements of the
reference semantics (aggregate.py / join.py / call.py)."""
import numpy as np
import pandas as pd
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def tpch():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from dask_sql_amd.context import Context
    rng = np.random.default_rng(7)
    n_o, n_l = 20_000, 80_000
    orders = pd.DataFrame({
        "o_orderkey": np.arange(n_o, dtype=np.int64),
        "o_orderdate": pd.to_datetime("1993-01-01")
        + pd.to_timedelta(rng.integers(0, 2000, n_o), unit="D"),
        "o_orderpriority": pd.Series(
            rng.choice(["1-URGENT", "2-HIGH", "3-MEDIUM", "4-NOT SPECIFIED",
                        "5-LOW"], n_o)).astype("category"),
    })
    li = pd.DataFrame({
        "l_orderkey": rng.integers(0, n_o, n_l).astype(np.int64),
        "l_partkey": rng.integers(0, 2000, n_l).astype(np.int64),
        "l_shipdate": pd.to_datetime("1993-01-01")
        + pd.to_timedelta(rng.integers(0, 2000, n_l), unit="D"),
        "l_commitdate": pd.to_datetime("1993-01-01")
        + pd.to_timedelta(rng.integers(0, 2000, n_l), unit="D"),
        "l_receiptdate": pd.to_datetime("1993-01-01")
        + pd.to_timedelta(rng.integers(0, 2000, n_l), unit="D"),
        "l_extendedprice": np.round(rng.random(n_l) * 1000, 2),
        "l_discount": np.round(rng.integers(0, 11, n_l) * 0.01, 2),
        "l_quantity": rng.integers(1, 51, n_l).astype(np.int64),
        "l_shipmode": pd.Series(rng.choice(
            ["MAIL", "SHIP", "AIR", "TRUCK"], n_l)).astype("category"),
    })
    part = pd.DataFrame({
        "p_partkey": np.arange(2000, dtype=np.int64),
        "p_type": pd.Series(rng.choice(
            ["PROMO BRUSHED", "STANDARD POLISHED", "PROMO PLATED",
             "ECONOMY BURNISHED"], 2000)).astype("category"),
    })
    c = Context()
    c.create_table("orders", orders)
    c.create_table("lineitem", li)
    c.create_table("part", part)
    return c, orders, li, part


def test_q6_forecast_revenue(tpch):
    c, orders, li, part = tpch
    got = c.sql(
        "SELECT SUM(l_extendedprice * l_discount) AS revenue FROM lineitem "
        "WHERE l_shipdate >= DATE '1994-01-01' "
        "AND l_shipdate < DATE '1994-01-01' + INTERVAL '1' YEAR "
        "AND l_discount BETWEEN 0.05 AND 0.07 AND l_quantity < 24").compute()
    m = ((li.l_shipdate >= "1994-01-01") & (li.l_shipdate < "1995-01-01")
         & (li.l_discount >= 0.05) & (li.l_discount <= 0.07)
         & (li.l_quantity < 24))
    exp = (li[m].l_extendedprice * li[m].l_discount).sum()
    np.testing.assert_allclose(float(got["revenue"].iloc[0]), exp, rtol=1e-9)


def test_q4_order_priority(tpch):
    c, orders, li, part = tpch
    got = c.sql(
        "SELECT o_orderpriority, COUNT(*) AS order_count FROM orders o "
        "WHERE o.o_orderdate >= DATE '1994-07-01' "
        "AND o.o_orderdate < DATE '1994-07-01' + INTERVAL '3' MONTH "
        "AND EXISTS (SELECT 1 FROM lineitem l WHERE "
        "l.l_orderkey = o.o_orderkey AND l.l_commitdate < l.l_receiptdate) "
        "GROUP BY o_orderpriority ORDER BY o_orderpriority").compute()
    late = set(li[li.l_commitdate < li.l_receiptdate].l_orderkey)
    om = orders[(orders.o_orderdate >= "1994-07-01")
                & (orders.o_orderdate < "1994-10-01")
                & orders.o_orderkey.isin(late)]
    exp = om.groupby("o_orderpriority", observed=True).size().sort_index()
    assert got["o_orderpriority"].tolist() == list(exp.index)
    assert got["order_count"].astype(int).tolist() == exp.tolist()


def test_q12_shipmode(tpch):
    c, orders, li, part = tpch
    got = c.sql(
        "SELECT l_shipmode, "
        "SUM(CASE WHEN o_orderpriority = '1-URGENT' "
        "OR o_orderpriority = '2-HIGH' THEN 1 ELSE 0 END) AS high_line, "
        "COUNT(*) AS total FROM orders, lineitem "
        "WHERE o_orderkey = l_orderkey AND l_shipmode IN ('MAIL', 'SHIP') "
        "AND l_receiptdate >= DATE '1994-01-01' "
        "GROUP BY l_shipmode ORDER BY l_shipmode").compute()
    j = li.merge(orders, left_on="l_orderkey", right_on="o_orderkey")
    j = j[j.l_shipmode.isin(["MAIL", "SHIP"])
          & (j.l_receiptdate >= "1994-01-01")]
    j["high"] = j.o_orderpriority.isin(["1-URGENT", "2-HIGH"]).astype(int)
    exp = j.groupby("l_shipmode", observed=True).agg(
        high_line=("high", "sum"), total=("high", "size")).sort_index()
    assert got["l_shipmode"].tolist() == list(exp.index)
    assert got["high_line"].astype(int).tolist() == exp.high_line.tolist()
    assert got["total"].astype(int).tolist() == exp.total.tolist()


def test_q14_promo_effect(tpch):
    c, orders, li, part = tpch
    got = c.sql(
        "SELECT 100.00 * SUM(CASE WHEN p_type LIKE 'PROMO%' "
        "THEN l_extendedprice * (1 - l_discount) ELSE 0.0 END) / "
        "SUM(l_extendedprice * (1 - l_discount)) AS promo_revenue "
        "FROM lineitem, part WHERE l_partkey = p_partkey "
        "AND l_shipdate >= DATE '1995-09-01' "
        "AND l_shipdate < DATE '1995-09-01' + INTERVAL '1' MONTH").compute()
    j = li.merge(part, left_on="l_partkey", right_on="p_partkey")
    j = j[(j.l_shipdate >= "1995-09-01") & (j.l_shipdate < "1995-10-01")]
    rev = j.l_extendedprice * (1 - j.l_discount)
    promo = rev.where(j.p_type.astype(str).str.startswith("PROMO"),
                      0.0).sum()
    exp = 100.0 * promo / rev.sum()
    np.testing.assert_allclose(float(got["promo_revenue"].iloc[0]), exp,
                               rtol=1e-9)
