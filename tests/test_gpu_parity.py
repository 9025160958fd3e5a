"""GPU parity suite (-m gpu): the product path (Context.sql → HIP kernels via
the C ABI) against the oracle and the reference's golden vectors
(SURVEY.md §8c). Bar: bit-exact for COUNT/integer/keys; ≤1e-6 relative for
fp64 SUM/AVG (BASELINE.md parity gate)."""
import numpy as np
import pandas as pd
import pytest

from tests.conftest import assert_frame_close, golden_expected

pytestmark = pytest.mark.gpu

REL_TOL = 1e-6


@pytest.fixture(scope="module")
def ctx(request):
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from dask_sql_amd.context import Context
    return Context()


@pytest.fixture(scope="module")
def c(ctx):
    from tests.conftest import golden_fixture_df
    ctx.create_table("user_table_1", golden_fixture_df("user_table_1"))
    ctx.create_table("user_table_2", golden_fixture_df("user_table_2"))
    ctx.create_table("df_simple", golden_fixture_df("df_simple"))
    ctx.create_table("user_table_nan", golden_fixture_df("user_table_nan"))
    np.random.seed(42)
    df700 = pd.DataFrame(
        {"a": [1.0] * 100 + [2.0] * 200 + [3.0] * 400,
         "b": 10 * np.random.rand(700)}
    )
    ctx.create_table("df", df700)
    return ctx


# ---------------------------------------------------------------- golden
def test_groupby_simple(c):
    out = c.sql('SELECT user_id, SUM(b) AS "S" FROM user_table_1 '
                "GROUP BY user_id").compute()
    out = out.sort_values("user_id").reset_index(drop=True)
    assert_frame_close(out, golden_expected("groupby_simple"))


def test_groupby_all(c):
    out = c.sql('SELECT SUM(b) AS "S", SUM(2) AS "X" FROM user_table_1'
                ).compute()
    assert_frame_close(out, golden_expected("groupby_all"))


def test_groupby_filtered_full(c):
    out = c.sql('SELECT SUM(b) FILTER (WHERE user_id = 2) AS "S1", '
                'SUM(b) AS "S2" FROM user_table_1').compute()
    assert_frame_close(out, golden_expected("groupby_filtered_full"))


def test_groupby_filtered_grouped(c):
    out = c.sql('SELECT user_id, SUM(b) FILTER (WHERE user_id = 2) AS "S1", '
                'SUM(b) AS "S2" FROM user_table_1 GROUP BY user_id').compute()
    out = out.sort_values("user_id").reset_index(drop=True)
    assert_frame_close(out, golden_expected("groupby_filtered_grouped"))


def test_join_inner(c):
    out = c.sql("SELECT lhs.user_id, lhs.b, rhs.c FROM user_table_1 AS lhs "
                "JOIN user_table_2 AS rhs ON lhs.user_id = rhs.user_id"
                ).compute()
    assert_frame_close(out, golden_expected("join_inner"),
                       sort_by=["user_id", "b", "c"])


def test_join_outer(c):
    out = c.sql("SELECT lhs.user_id, lhs.b, rhs.c FROM user_table_1 AS lhs "
                "FULL JOIN user_table_2 AS rhs ON lhs.user_id = rhs.user_id"
                ).compute()
    assert_frame_close(out, golden_expected("join_outer"),
                       sort_by=["user_id", "b", "c"])


def test_join_left(c):
    out = c.sql("SELECT lhs.user_id, lhs.b, rhs.c FROM user_table_1 AS lhs "
                "LEFT JOIN user_table_2 AS rhs ON lhs.user_id = rhs.user_id"
                ).compute()
    assert_frame_close(out, golden_expected("join_left"),
                       sort_by=["user_id", "b", "c"])


def test_join_residual(c):
    out = c.sql("SELECT lhs.user_id, lhs.b, rhs.user_id, rhs.c "
                "FROM user_table_1 AS lhs JOIN user_table_2 AS rhs "
                "ON rhs.user_id = lhs.user_id AND rhs.c - lhs.b >= 0"
                ).compute()
    assert_frame_close(out, golden_expected("join_residual"),
                       sort_by=["lhs.user_id", "b", "c"])


def test_join_conditional(c):
    out = c.sql("SELECT lhs.a, lhs.b, rhs.a, rhs.b FROM df_simple AS lhs "
                "JOIN df_simple AS rhs ON lhs.a < rhs.b AND lhs.b < rhs.a"
                ).compute()
    assert_frame_close(out, golden_expected("join_conditional"),
                       sort_by=["lhs.a", "rhs.a"])


def test_join_literal_true(c):
    out = c.sql("SELECT lhs.user_id, lhs.b, rhs.user_id, rhs.c "
                "FROM user_table_1 AS lhs JOIN user_table_2 AS rhs ON True"
                ).compute()
    assert_frame_close(
        out, golden_expected("join_literal_true"),
        sort_by=["lhs.user_id", "b", "rhs.user_id", "c"])


def test_join_literal_false(c):
    out = c.sql("SELECT lhs.user_id, lhs.b, rhs.user_id, rhs.c "
                "FROM user_table_1 AS lhs JOIN user_table_2 AS rhs ON False"
                ).compute()
    assert len(out) == 0


def test_filter_with_nan(c):
    out = c.sql("SELECT * FROM user_table_nan WHERE c = 3").compute()
    assert out["c"].astype(np.int64).tolist() == [3]


def test_filter_simple(c):
    out = c.sql("SELECT * FROM df WHERE a < 2").compute()
    np.random.seed(42)
    b = 10 * np.random.rand(700)
    exp = pd.DataFrame({"a": [1.0] * 100, "b": b[:100]})
    assert_frame_close(out, exp)  # order-preserving compaction


def test_filter_scalar(c):
    # reference test_filter.py:20-39 (filter_or_scalar short-circuit)
    assert len(c.sql("SELECT * FROM df WHERE True").compute()) == 700
    assert len(c.sql("SELECT * FROM df WHERE False").compute()) == 0
    assert len(c.sql("SELECT * FROM df WHERE (1 = 1)").compute()) == 700
    assert len(c.sql("SELECT * FROM df WHERE (1 = 0)").compute()) == 0


def test_distinct(c):
    out = c.sql("SELECT DISTINCT user_id FROM user_table_1").compute()
    assert sorted(out["user_id"].tolist()) == [1, 2, 3]


def test_sum_distinct(c):
    out = c.sql("SELECT user_id, SUM(DISTINCT b) AS s FROM user_table_1 "
                "GROUP BY user_id").compute()
    out = out.sort_values("user_id").reset_index(drop=True)
    # user 2 has b = {3, 1} → 4; user 1 b={3} → 3; user 3 → 3
    assert out["s"].astype(np.int64).tolist() == [3, 4, 3]


def test_groupby_null_group(ctx):
    # aggregate.py:575-577 dropna=False: NULL group kept
    ctx.create_table("nullkeys", pd.DataFrame({
        "k": pd.array([1, None, None, 2], dtype="Int64"),
        "v": [1.0, 2.0, 3.0, 4.0]}))
    out = ctx.sql("SELECT k, SUM(v) AS s FROM nullkeys GROUP BY k").compute()
    assert len(out) == 3
    nullrow = out[out["k"].isna()]
    assert len(nullrow) == 1 and abs(nullrow["s"].iloc[0] - 5.0) < 1e-12


def test_sum_all_null_is_null(ctx):
    # custom_sum min_count=1 (aggregate.py:486-493)
    ctx.create_table("nullvals", pd.DataFrame({
        "k": [1, 1, 2], "v": pd.array([None, None, 5], dtype="Float64")}))
    out = ctx.sql("SELECT k, SUM(v) AS s FROM nullvals GROUP BY k").compute()
    out = out.sort_values("k").reset_index(drop=True)
    assert np.isnan(out["s"][0]) and out["s"][1] == 5.0


def test_orderby_limit(c):
    out = c.sql('SELECT user_id, SUM(b) AS "S" FROM user_table_1 '
                'GROUP BY user_id ORDER BY "S" DESC, user_id LIMIT 2'
                ).compute()
    assert out["user_id"].tolist() == [2, 1]
    assert out["S"].astype(np.int64).tolist() == [4, 3]


def test_case_when(c):
    out = c.sql("SELECT user_id, CASE WHEN b > 2 THEN 1 ELSE 0 END AS x "
                "FROM user_table_1").compute()
    assert sorted(out["x"].astype(np.int64).tolist()) == [0, 1, 1, 1]


def test_strings_decode(ctx):
    ctx.create_table("stab", pd.DataFrame({
        "s": ["BUILDING", "AUTO", "BUILDING"], "v": [1, 2, 3]}))
    out = ctx.sql("SELECT s, SUM(v) AS t FROM stab WHERE s = 'BUILDING' "
                  "GROUP BY s").compute()
    assert out["s"].tolist() == ["BUILDING"]
    assert out["t"].astype(np.int64).tolist() == [4]


# ------------------------------------------------------- oracle parity
def _oracle_ctx(ctx, name, pdf, **kw):
    ctx.create_table(name, pdf, **kw)


def test_c1_parity(ctx):
    from datagen import gen_c1
    from oracle.tpch import oracle_c1_c2_groupby
    key, x = gen_c1(n=1_000_000, n_groups=1_000)
    ctx.create_table("c1", pd.DataFrame({"key": key, "x": x}))
    out = ctx.sql("SELECT key, SUM(x) AS s, COUNT(*) AS c FROM c1 "
                  "GROUP BY key").compute()
    out = out.sort_values("key").reset_index(drop=True)
    exp = oracle_c1_c2_groupby(key, x, predicate=False)
    assert (out["key"].to_numpy() == exp["key"].to_numpy()).all()
    assert (out["c"].to_numpy() == exp["c"].to_numpy()).all()  # bit-exact
    assert np.allclose(out["s"], exp["s"], rtol=REL_TOL)


def test_c2_parity(ctx):
    from datagen import gen_c2
    from oracle.tpch import oracle_c1_c2_groupby
    key, val = gen_c2(n=2_000_000, n_groups=100_000)  # global-table path
    ctx.create_table("c2", pd.DataFrame({"key": key, "x": val}))
    out = ctx.sql("SELECT key, SUM(x) AS s, COUNT(*) AS c FROM c2 "
                  "WHERE x < 0.5 GROUP BY key").compute()
    out = out.sort_values("key").reset_index(drop=True)
    exp = oracle_c1_c2_groupby(key, val, predicate=True)
    assert (out["key"].to_numpy() == exp["key"].to_numpy()).all()
    assert (out["c"].to_numpy() == exp["c"].to_numpy()).all()
    assert np.allclose(out["s"], exp["s"], rtol=REL_TOL)


def test_c3_parity(ctx):
    from datagen import gen_c3
    from oracle.tpch import oracle_c3_join
    bk, bv, pk, pv = gen_c3(n_build=100_000, n_probe=1_000_000)
    ctx.create_table("probe_t", pd.DataFrame({"key": pk, "pv": pv}))
    ctx.create_table("build_t", pd.DataFrame({"key": bk, "bv": bv}))
    out = ctx.sql("SELECT p.key, p.pv, b.bv FROM probe_t p JOIN build_t b "
                  "ON p.key = b.key").compute()
    exp = oracle_c3_join(bk, pk, bv, pv)
    # unordered output: sort-normalize both
    out = out.sort_values(["key", "pv"]).reset_index(drop=True)
    exp = exp.sort_values(["key", "pv"]).reset_index(drop=True)
    assert len(out) == len(exp)
    assert (out["key"].to_numpy() == exp["key"].to_numpy()).all()
    assert np.allclose(out["pv"], exp["pv"], rtol=REL_TOL)
    assert np.allclose(out["bv"], exp["bv"], rtol=REL_TOL)


Q1_SQL = """
SELECT l_returnflag, l_linestatus, SUM(l_quantity) AS sum_qty,
       SUM(l_extendedprice) AS sum_base_price,
       SUM(l_extendedprice*(1-l_discount)) AS sum_disc_price,
       SUM(l_extendedprice*(1-l_discount)*(1+l_tax)) AS sum_charge,
       AVG(l_quantity) AS avg_qty, AVG(l_extendedprice) AS avg_price,
       AVG(l_discount) AS avg_disc, COUNT(*) AS count_order
FROM lineitem WHERE l_shipdate <= 10471
GROUP BY l_returnflag, l_linestatus
ORDER BY l_returnflag, l_linestatus
"""


def test_q1_parity(ctx):
    from datagen import gen_lineitem_q1
    from oracle.tpch import oracle_q1
    li = gen_lineitem_q1(n=1_000_000)
    ctx.create_table("lineitem", li)  # shipdate as plain int32 (day ints)
    out = ctx.sql(Q1_SQL).compute()
    exp = oracle_q1(li)
    assert (out["l_returnflag"].to_numpy().astype(np.int64)
            == exp["l_returnflag"].to_numpy().astype(np.int64)).all()
    assert (out["count_order"].to_numpy().astype(np.int64)
            == exp["count_order"].to_numpy()).all()
    for col in ("sum_qty", "sum_base_price", "sum_disc_price", "sum_charge",
                "avg_qty", "avg_price", "avg_disc"):
        assert np.allclose(out[col], exp[col], rtol=REL_TOL), col


Q3_SQL = """
SELECT l_orderkey, SUM(l_extendedprice*(1-l_discount)) AS revenue,
       o_orderdate, o_shippriority
FROM customer, orders, lineitem3
WHERE c_mktsegment = 0 AND c_custkey = o_custkey
  AND l_orderkey = o_orderkey AND o_orderdate < 9204
  AND l_shipdate > 9204
GROUP BY l_orderkey, o_orderdate, o_shippriority
ORDER BY revenue DESC, o_orderdate LIMIT 10
"""


def test_q3_parity(ctx):
    from datagen import gen_q3
    from oracle.tpch import oracle_q3
    cust, orders, li = gen_q3(sf_rows=(20_000, 100_000, 400_000))
    ctx.create_table("customer", cust)
    ctx.create_table("orders", orders)
    ctx.create_table("lineitem3", li)
    out = ctx.sql(Q3_SQL).compute()
    exp = oracle_q3(cust, orders, li)
    assert len(out) == len(exp)
    assert (out["l_orderkey"].to_numpy().astype(np.int64)
            == exp["l_orderkey"].to_numpy()).all()
    assert np.allclose(out["revenue"], exp["revenue"], rtol=REL_TOL)
    assert (out["o_orderdate"].to_numpy().astype(np.int64)
            == exp["o_orderdate"].to_numpy()).all()


# ---------------------------------------------- full-size properties
def test_c2_properties_large(ctx):
    """Size-independent invariants at a larger scale (global hash path):
    count conservation, key-set equality, checksum of sums vs numpy."""
    from datagen import gen_c2
    n, g = 20_000_000, 1_000_000
    key, val = gen_c2(n=n, n_groups=g)
    ctx.create_table("c2big", pd.DataFrame({"key": key, "x": val}))
    out = ctx.sql("SELECT key, SUM(x) AS s, COUNT(*) AS c FROM c2big "
                  "WHERE x < 0.5 GROUP BY key").compute()
    sel = val < 0.5
    assert out["c"].sum() == sel.sum()  # bit-exact count conservation
    np_keys = np.unique(key[sel])
    got_keys = np.sort(out["key"].to_numpy())
    assert len(got_keys) == len(np_keys) and (got_keys == np_keys).all()
    assert np.isclose(out["s"].sum(), val[sel].sum(), rtol=1e-9)


def test_filter_order_preserving_direct(ctx):
    """dsx_filter emits row ids in ascending row order (pandas boolean-mask
    take preserves order, filter.py:40)."""
    rtm = ctx._get_runtime()
    from dask_sql_amd import runtime as rt
    rng = np.random.default_rng(7)
    x = rng.random(3_000_000)
    col = rtm.upload_column(x)
    prog = rtm.make_prog([(1, 0, 0), (2, 0, 0.25), (20, 0, 0)])  # x < 0.25
    sel_ptr, count = rtm.filter(prog, [col], len(x))
    sel = rtm.wrap_sel(sel_ptr, count)
    ids = np.empty(count, dtype=np.uint32)
    rtm._download(sel.data, ids)
    exp = np.nonzero(x < 0.25)[0]
    assert count == len(exp)
    assert (ids.astype(np.int64) == exp).all()  # exact order


def test_partition_stable_direct(ctx):
    """dsx_partition: bucket-contiguous, stable within bucket."""
    rtm = ctx._get_runtime()
    rng = np.random.default_rng(11)
    codes = rng.integers(0, 1000, size=1_000_000).astype(np.int64)
    col = rtm.upload_column(codes)
    sel, offsets = rtm.partition(col, 8)
    ids = np.empty(len(codes), dtype=np.uint32)
    rtm._download(sel.data, ids)
    assert offsets[-1] == len(codes)
    # recompute expected bucket of each row (mix64 of code ^ salt mod 8)
    import ctypes
    got_buckets = np.zeros(len(codes), dtype=np.int64)
    for b in range(8):
        got_buckets[ids[offsets[b]:offsets[b + 1]].astype(np.int64)] = b
    # stability: within each bucket ids ascend
    for b in range(8):
        part = ids[offsets[b]:offsets[b + 1]].astype(np.int64)
        assert (np.diff(part) > 0).all()
    # same code → same bucket
    for code in rng.choice(1000, 10):
        rows = np.nonzero(codes == code)[0]
        assert len(np.unique(got_buckets[rows])) == 1


def test_filter_idle_block_word_alias(ctx):
    """Regression: idle grid blocks (lo clamped to n) re-counted the last
    partial mask word when n % 64 != 0 — filter count came out high by
    popcount(last word) × idle blocks (found at Q3's customer scan)."""
    rtm = ctx._get_runtime()
    rng = np.random.default_rng(5)
    seg = rng.integers(0, 5, 1_500_000).astype(np.int8)  # n % 64 == 32
    col = rtm.upload_column(seg)
    prog = rtm.make_prog([(1, 0, 0), (3, 0, 0), (34, 0, 0)])  # seg == 0
    sel_ptr, count = rtm.filter(prog, [col], len(seg))
    sel = rtm.wrap_sel(sel_ptr, count)
    ids = np.empty(count, dtype=np.uint32)
    rtm._download(sel.data, ids)
    exp = np.nonzero(seg == 0)[0]
    assert count == len(exp)
    assert (ids.astype(np.int64) == exp).all()


def test_q3_real_text_parity(ctx):
    """The bench headline path: REAL Q3 text ('BUILDING',
    DATE '1995-03-15') over dictionary/DATE-typed tables must match the
    oracle (which computes on codes/day-ints — monotone-equivalent,
    SURVEY §8 appendix)."""
    from datagen import Q3_SQL, gen_q3, register_q3_tables
    from dask_sql_amd.context import Context
    from oracle.tpch import oracle_q3
    cust, orders, li = gen_q3(sf_rows=(20_000, 100_000, 400_000))
    c = Context()
    register_q3_tables(c, cust, orders, li)
    out = c.sql(Q3_SQL).compute()
    exp = oracle_q3(cust, orders, li)
    assert len(out) == len(exp)
    assert (out["l_orderkey"].to_numpy().astype(np.int64)
            == exp["l_orderkey"].to_numpy()).all()
    assert np.allclose(out["revenue"], exp["revenue"], rtol=REL_TOL)
    # o_orderdate comes back as datetime64 (DATE column); oracle holds
    # day-ints
    got_days = out["o_orderdate"].to_numpy().astype(
        "datetime64[D]").astype(np.int64)
    assert (got_days == exp["o_orderdate"].to_numpy()).all()


def test_q1_real_text_parity(ctx):
    """Q1 with reference literals (DATE - INTERVAL folding) over
    dictionary returnflag/linestatus and DATE shipdate."""
    from datagen import (Q1_SQL, RETURNFLAG_DICT, LINESTATUS_DICT,
                         gen_lineitem_q1, register_q1_table)
    from dask_sql_amd.context import Context
    from oracle.tpch import oracle_q1
    li = gen_lineitem_q1(n=500_000, seed=5)
    c = Context()
    register_q1_table(c, li)
    out = c.sql(Q1_SQL).compute()
    exp = oracle_q1(li)  # keys are codes, sorted
    out = out.assign(
        rf=out["l_returnflag"].map(
            {s: i for i, s in enumerate(RETURNFLAG_DICT)}),
        ls=out["l_linestatus"].map(
            {s: i for i, s in enumerate(LINESTATUS_DICT)}),
    ).sort_values(["rf", "ls"]).reset_index(drop=True)
    assert out["rf"].tolist() == exp["l_returnflag"].tolist()
    assert out["ls"].tolist() == exp["l_linestatus"].tolist()
    assert (out["count_order"].to_numpy().astype(np.int64)
            == exp["count_order"].to_numpy()).all()
    for col in ["sum_qty", "sum_base_price", "sum_disc_price", "sum_charge",
                "avg_qty", "avg_price", "avg_disc"]:
        assert np.allclose(out[col], exp[col], rtol=REL_TOL), col


def test_group_by_case_reference(c):
    """reference test_groupby.py:155-171: expression GROUP BY key
    (user_id + 1) with SUM(CASE WHEN b = 3 THEN 1 END) — expected frame
    A=[2,3,4], S=[1,1,1]."""
    out = c.sql('SELECT user_id + 1 AS "A", '
                'SUM(CASE WHEN b = 3 THEN 1 END) AS "S" '
                "FROM user_table_1 GROUP BY user_id + 1").compute()
    out = out.sort_values("A").reset_index(drop=True)
    assert out["A"].astype(np.int64).tolist() == [2, 3, 4]
    assert out["S"].astype(np.float64).tolist() == [1.0, 1.0, 1.0]
