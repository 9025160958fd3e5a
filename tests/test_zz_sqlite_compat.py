"""SQLite-differential compatibility suite (-m gpu): the reference runs the
fugue-sql compatibility corpus against sqlite3
(tests/integration/test_compatibility.py) — we run the same SQL fixtures
through OUR engine and through sqlite3 (stdlib, in-memory) and compare.
Data-driven: one table of (name, SQL, frame specs); frames are rebuilt with
the reference's make_rand_df recipe (np.random.seed(0) per frame). Ordered
queries compare row-by-row; unordered compare canonically sorted."""
import sqlite3
from datetime import datetime, timedelta

import numpy as np
import pandas as pd
import pytest

from tests.test_gpu_semantics import ctx  # noqa: F401

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(180)]
# the timeout bounds a hang in these UNVALIDATED paths to a test failure
# instead of a dead GPU box (pytest-timeout is in the image)


def rand_df(size, **kwargs):
    """The reference's make_rand_df recipe (test_compatibility.py:49-84):
    seeded per call, value pools of 10, explicit null counts."""
    np.random.seed(0)
    data = {}
    for k, v in kwargs.items():
        if not isinstance(v, tuple):
            v = (v, 0.0)
        dt, null_ct = v[0], v[1]
        if dt is int:
            s = np.random.randint(10, size=size)
        elif dt is bool:
            s = np.where(np.random.randint(2, size=size), True, False)
        elif dt is float:
            s = np.random.rand(size)
        elif dt is str:
            pool = [f"ssssss{x}" for x in range(10)]
            s = np.array([pool[x]
                          for x in np.random.randint(10, size=size)])
        elif dt is datetime:
            pool = [datetime(2020, 1, 1) + timedelta(days=x)
                    for x in range(10)]
            s = np.array([pool[x]
                          for x in np.random.randint(10, size=size)])
        else:
            raise NotImplementedError(dt)
        ps = pd.Series(s)
        if null_ct > 0:
            idx = np.random.choice(size, null_ct, replace=False).tolist()
            ps[idx] = None
        data[k] = ps
    return pd.DataFrame(data)


_n = [0]
PLAN_ONLY = False  # tests/test_planner.py flips this to reuse the corpus
                   # as a CPU planning sweep


def eq_sqlite(ctx, sql, **dfs):
    _n[0] += 1
    for name, df in dfs.items():
        reg = df.copy()
        for c in reg.columns:
            if reg[c].dtype == object and reg[c].map(
                    lambda x: isinstance(x, str) or x is None).all():
                reg[c] = reg[c].astype("category")
        ctx.create_table(name, reg)  # plain names, re-registered per case
    if PLAN_ONLY:
        ctx._get_ral(sql)
        return
    engine = sqlite3.connect(":memory:")
    for name, df in dfs.items():
        df.to_sql(name, engine, index=False)
    ours = ctx.sql(sql).compute()
    theirs = pd.read_sql(sql, engine)
    engine.close()
    _compare(ours, theirs, ordered="ORDER BY" in sql.upper())


def _norm_col(s):
    if s.dtype == object:
        try:
            return pd.to_numeric(s)
        except Exception:
            return s.astype(str).where(s.notna(), None)
    if str(s.dtype).startswith("datetime"):
        return s
    return s


def _compare(ours, theirs, ordered):
    assert [c.lower() for c in ours.columns] == \
        [c.lower() for c in theirs.columns], (list(ours.columns),
                                              list(theirs.columns))
    assert len(ours) == len(theirs), (len(ours), len(theirs))
    a = ours.reset_index(drop=True)
    b = theirs.reset_index(drop=True)
    a.columns = [c.lower() for c in a.columns]
    b.columns = [c.lower() for c in b.columns]
    for c in b.columns:
        if str(a[c].dtype).startswith("datetime"):
            b[c] = pd.to_datetime(b[c])
    if not ordered:
        key = [repr(tuple(None if pd.isna(v) else
                          (round(v, 6) if isinstance(v, float) else v)
                          for v in row))
               for row in a.itertuples(index=False)]
        a = a.iloc[np.argsort(key, kind="stable")].reset_index(drop=True)
        key = [repr(tuple(None if pd.isna(v) else
                          (round(v, 6) if isinstance(v, float) else v)
                          for v in row))
               for row in b.itertuples(index=False)]
        b = b.iloc[np.argsort(key, kind="stable")].reset_index(drop=True)
    for c in b.columns:
        av, bv = _norm_col(a[c]), _norm_col(b[c])
        try:
            avn = av.astype(float)
            bvn = bv.astype(float)
            both = ~(avn.isna() & bvn.isna())
            assert np.allclose(avn[both], bvn[both], rtol=1e-9,
                               equal_nan=True), c
        except (ValueError, TypeError):
            assert [None if pd.isna(x) else str(x) for x in av] == \
                [None if pd.isna(x) else str(x) for x in bv], c


# ---- the corpus (SQL fixtures from test_compatibility.py) ------------------
def test_sqlc_basic_select(ctx):
    df = rand_df(5, a=(int, 2), b=(str, 3), c=(float, 4))
    eq_sqlite(ctx, "SELECT 1 AS a, 1.5 AS b, 'x' AS c")
    eq_sqlite(ctx, "SELECT 1+2 AS a, 1.5*3 AS b, 'x' AS c")
    eq_sqlite(ctx, "SELECT * FROM a", a=df)
    eq_sqlite(ctx, "SELECT * FROM a AS x", a=df)
    eq_sqlite(ctx, "SELECT b AS bb, a+1-2*3.0/4 AS cc FROM a", a=df)
    eq_sqlite(ctx, "SELECT b AS bb, a+1-2*3.0/4 AS cc, x.* FROM a AS x",
              a=df)


def test_sqlc_case_when(ctx):
    a = rand_df(100, a=(int, 20), b=(str, 30), c=(float, 40))
    eq_sqlite(ctx, """
        SELECT a,b,c,
            CASE WHEN a<10 THEN a+3 WHEN c<0.5 THEN a+5
                 ELSE (1+2)*3 + a END AS d
        FROM a""", a=a)


def test_sqlc_drop_duplicates(ctx):
    a = rand_df(100, a=int, b=int)
    eq_sqlite(ctx, "SELECT DISTINCT b, a FROM a "
                   "ORDER BY a NULLS LAST, b NULLS FIRST", a=a)
    a2 = rand_df(100, a=(int, 50), b=(int, 50))
    eq_sqlite(ctx, "SELECT DISTINCT b, a FROM a "
                   "ORDER BY a NULLS LAST, b NULLS FIRST", a=a2)
    a3 = rand_df(100, a=(int, 50), b=(str, 50), c=float)
    eq_sqlite(ctx, "SELECT DISTINCT b, a FROM a "
                   "ORDER BY a NULLS LAST, b NULLS FIRST", a=a3)


def test_sqlc_order_by(ctx):
    a = rand_df(100, a=(int, 20), b=(str, 30))
    eq_sqlite(ctx, "SELECT * FROM a ORDER BY a NULLS FIRST, "
                   "b NULLS LAST", a=a)
    eq_sqlite(ctx, "SELECT * FROM a ORDER BY a NULLS LAST, "
                   "b NULLS FIRST LIMIT 20", a=a)
    b = rand_df(100, a=(float, 20), b=(str, 30))
    eq_sqlite(ctx, "SELECT * FROM a ORDER BY a DESC NULLS FIRST, "
                   "b NULLS LAST LIMIT 15", a=b)


def test_sqlc_where(ctx):
    df = rand_df(100, a=(int, 30), b=(str, 30), c=(float, 30))
    eq_sqlite(ctx, "SELECT * FROM a WHERE a<5 AND b IS NOT NULL", a=df)
    eq_sqlite(ctx, "SELECT * FROM a WHERE a<5 OR b IS NULL", a=df)
    eq_sqlite(ctx, "SELECT * FROM a WHERE c IS NOT NULL", a=df)


def test_sqlc_in_between(ctx):
    df = rand_df(10, a=(int, 3), b=(str, 3))
    eq_sqlite(ctx, "SELECT * FROM a WHERE a IN (2,4,6)", a=df)
    eq_sqlite(ctx, "SELECT * FROM a WHERE a BETWEEN 2 AND 4", a=df)
    eq_sqlite(ctx, "SELECT * FROM a WHERE a NOT IN (2,4,6) "
                   "AND a IS NOT NULL", a=df)
    eq_sqlite(ctx, "SELECT * FROM a WHERE a NOT BETWEEN 2 AND 4 "
                   "AND a IS NOT NULL", a=df)


def test_sqlc_joins(ctx):
    a = rand_df(100, a=int, b=(str, 50))
    b = rand_df(80, a=int, c=(str, 50))
    eq_sqlite(ctx, "SELECT a.a, a.b, c FROM a INNER JOIN b "
                   "ON a.a = b.a", a=a, b=b)
    eq_sqlite(ctx, "SELECT a.a, a.b, c FROM a LEFT JOIN b "
                   "ON a.a = b.a", a=a, b=b)
    x = rand_df(10, a=int)
    y = rand_df(20, b=int)
    eq_sqlite(ctx, "SELECT * FROM x CROSS JOIN y", x=x, y=y)


def test_sqlc_agg_group_by(ctx):
    a = rand_df(100, a=int, b=(str, 50), c=(int, 30), d=(str, 40),
                e=(float, 40))
    eq_sqlite(ctx, "SELECT a, b, COUNT(c) AS c_cnt, SUM(c) AS c_sum, "
                   "AVG(e) AS e_avg, MIN(c) AS c_min, MAX(c) AS c_max "
                   "FROM a GROUP BY a, b", a=a)
    eq_sqlite(ctx, "SELECT a, COUNT(DISTINCT c) AS cd FROM a GROUP BY a",
              a=a)
    eq_sqlite(ctx, "SELECT COUNT(*) AS n, SUM(c) AS s, AVG(e) AS m FROM a",
              a=a)


def test_sqlc_window_row_number(ctx):
    a = rand_df(10, a=int, b=(float, 5))
    eq_sqlite(ctx, "SELECT *, ROW_NUMBER() OVER (ORDER BY a ASC, "
                   "b DESC NULLS LAST) AS x FROM a ORDER BY x", a=a)
    eq_sqlite(ctx, "SELECT b, ROW_NUMBER() OVER (PARTITION BY b ORDER "
                   "BY a) AS x FROM a WHERE b IS NOT NULL ORDER BY b, x",
              a=a)


def test_sqlc_window_ranks(ctx):
    a = rand_df(100, a=int, b=(float, 50), c=(str, 50))
    eq_sqlite(ctx, "SELECT a, b, RANK() OVER (ORDER BY a, b DESC "
                   "NULLS LAST) AS r, DENSE_RANK() OVER (ORDER BY a, "
                   "b DESC NULLS LAST) AS d FROM a ORDER BY a, "
                   "b DESC NULLS LAST", a=a)


def test_sqlc_window_lead_lag(ctx):
    a = rand_df(100, a=float, b=(int, 50), c=(str, 50))
    eq_sqlite(ctx, "SELECT LAG(b, 1) OVER (ORDER BY a) AS l1, "
                   "LEAD(b, 1) OVER (ORDER BY a) AS l2, a FROM a "
                   "ORDER BY a", a=a)


def test_sqlc_window_sum_frames(ctx):
    a = rand_df(100, a=float, b=(int, 50), c=(str, 50))
    eq_sqlite(ctx, "SELECT a, SUM(b) OVER (ORDER BY a ROWS BETWEEN "
                   "2 PRECEDING AND CURRENT ROW) AS s FROM a ORDER BY a",
              a=a)
    eq_sqlite(ctx, "SELECT a, SUM(b) OVER (ORDER BY a ROWS BETWEEN "
                   "UNBOUNDED PRECEDING AND CURRENT ROW) AS s FROM a "
                   "ORDER BY a", a=a)


def test_sqlc_nested_and_with(ctx):
    a = rand_df(100, a=int, b=(str, 50))
    eq_sqlite(ctx, "SELECT a, b FROM (SELECT a, b FROM a WHERE a >= 2) x "
                   "WHERE a < 7", a=a)
    eq_sqlite(ctx, "WITH x AS (SELECT a, b FROM a WHERE a > 2), "
                   "y AS (SELECT a, b FROM x WHERE a < 7) "
                   "SELECT * FROM y WHERE b IS NOT NULL", a=a)


def test_sqlc_set_ops(ctx):
    a = rand_df(30, a=(int, 10), b=(str, 10))
    b = rand_df(80, a=(int, 50), b=(str, 50))
    eq_sqlite(ctx, "SELECT * FROM a UNION SELECT * FROM b", a=a, b=b)
    eq_sqlite(ctx, "SELECT * FROM a UNION ALL SELECT * FROM b", a=a, b=b)
    eq_sqlite(ctx, "SELECT * FROM a EXCEPT SELECT * FROM b", a=a, b=b)
    eq_sqlite(ctx, "SELECT * FROM a INTERSECT SELECT * FROM b", a=a, b=b)


def test_sqlc_integration_1(ctx):
    # the reference's closing integration query (test_compatibility.py:
    # test_integration_1), CTE + window + HAVING composition
    a = rand_df(100, a=int, b=(str, 50), c=(int, 30))
    eq_sqlite(ctx, """
        WITH cte AS (SELECT a, b, COUNT(c) AS n FROM a
                     GROUP BY a, b HAVING COUNT(c) > 0)
        SELECT a, b, n, ROW_NUMBER() OVER (PARTITION BY b ORDER BY a,
            n) AS r
        FROM cte WHERE b IS NOT NULL ORDER BY b, r""", a=a)


def test_sqlc_window_min_max(ctx):
    for func in ["MIN", "MAX"]:
        a = rand_df(100, a=float, b=(int, 50), c=(str, 50))
        eq_sqlite(ctx, f"""
            SELECT a,b,
                {func}(b) OVER () AS a1,
                {func}(b) OVER (PARTITION BY c) AS a2,
                {func}(b+a) OVER (PARTITION BY c,b) AS a3,
                {func}(b+a) OVER (PARTITION BY b ORDER BY a
                    ROWS BETWEEN UNBOUNDED PRECEDING AND CURRENT ROW)
                    AS a4,
                {func}(b+a) OVER (PARTITION BY b ORDER BY a DESC
                    ROWS BETWEEN 2 PRECEDING AND CURRENT ROW) AS a5
            FROM a
            ORDER BY a NULLS FIRST, b NULLS FIRST, c NULLS FIRST""",
            a=a)
        eq_sqlite(ctx, f"""
            SELECT a,b,
                {func}(b) OVER (ORDER BY a DESC
                    ROWS BETWEEN 2 PRECEDING AND 1 PRECEDING) AS a6,
                {func}(b) OVER (ORDER BY a DESC
                    ROWS BETWEEN 2 PRECEDING AND 1 FOLLOWING) AS a7,
                {func}(b) OVER (ORDER BY a DESC
                    ROWS BETWEEN 2 PRECEDING AND UNBOUNDED FOLLOWING)
                    AS a8
            FROM a
            ORDER BY a NULLS FIRST, b NULLS FIRST, c NULLS FIRST""",
            a=a)


def test_sqlc_window_count_frames(ctx):
    a = rand_df(100, a=float, b=(int, 50), c=(str, 50))
    eq_sqlite(ctx, """
        SELECT a,b,
            COUNT(b) OVER (PARTITION BY c) AS a2,
            COUNT(b) OVER (PARTITION BY b ORDER BY a
                ROWS BETWEEN UNBOUNDED PRECEDING AND CURRENT ROW) AS a4,
            COUNT(b) OVER (ORDER BY a DESC
                ROWS BETWEEN 2 PRECEDING AND 1 FOLLOWING) AS a7
        FROM a
        ORDER BY a NULLS FIRST, b NULLS FIRST, c NULLS FIRST""", a=a)


def test_sqlc_window_sum_avg_partition(ctx):
    a = rand_df(100, a=float, b=(int, 50), c=(str, 50))
    eq_sqlite(ctx, """
        SELECT a,b,
            SUM(b) OVER () AS a1,
            AVG(b) OVER (PARTITION BY c) AS a2,
            SUM(b+a) OVER (PARTITION BY c,b) AS a3,
            AVG(b+a) OVER (PARTITION BY b ORDER BY a
                ROWS BETWEEN UNBOUNDED PRECEDING AND CURRENT ROW) AS a4
        FROM a
        ORDER BY a NULLS FIRST, b NULLS FIRST, c NULLS FIRST""", a=a)


def test_sqlc_multi_count_distinct(ctx):
    a = rand_df(100, a=(int, 50), b=(str, 50), c=(int, 30), d=(str, 40),
                e=(float, 40))
    eq_sqlite(ctx, """
        SELECT COUNT(a) AS c_a, COUNT(DISTINCT a) AS cd_a,
               COUNT(b) AS c_b, COUNT(DISTINCT b) AS cd_b,
               COUNT(c) AS c_c, COUNT(DISTINCT c) AS cd_c,
               COUNT(e) AS c_e
        FROM a""", a=a)
    eq_sqlite(ctx, """
        SELECT a, COUNT(DISTINCT c) AS cd_c, COUNT(DISTINCT d) AS cd_d
        FROM a GROUP BY a""", a=a)


def test_sqlc_agg_sum_avg_expr_group(ctx):
    a = rand_df(100, a=int, b=(str, 50), c=(int, 30), d=(str, 40),
                e=(float, 40))
    eq_sqlite(ctx, """
        SELECT a, b, a+1 AS c, SUM(c) AS sum_c, AVG(e) AS avg_e
        FROM a GROUP BY a, b""", a=a)
    eq_sqlite(ctx, "SELECT SUM(e) AS s, AVG(e) AS m FROM a", a=a)


def test_sqlc_float_count_distinct_last(ctx):
    # float DISTINCT aggregates ride the float-key groupby; defined last
    # so an execution gap here cannot mask the rest under pytest -x
    a = rand_df(100, e=(float, 40), a=int)
    eq_sqlite(ctx, "SELECT COUNT(DISTINCT e) AS cd_e FROM a", a=a)
    eq_sqlite(ctx, "SELECT a, COUNT(DISTINCT e) AS cd FROM a GROUP BY a",
              a=a)


def test_sqlc_agg_min_max(ctx):
    a = rand_df(100, a=(int, 50), b=(str, 50), c=(int, 30), d=(str, 40),
                e=(float, 40), g=(datetime, 40))
    eq_sqlite(ctx, """
        SELECT
            MIN(a) AS min_a, MAX(a) AS max_a,
            MIN(b) AS min_b, MAX(b) AS max_b,
            MIN(c) AS min_c, MAX(c) AS max_c,
            MIN(e) AS min_e, MAX(e) AS max_e,
            MIN(g) AS min_g, MAX(g) AS max_g,
            MIN(a+e) AS mix_1, MIN(a)+MIN(e) AS mix_2
        FROM a""", a=a)
    eq_sqlite(ctx, """
        SELECT a, b, a+1 AS c,
            MIN(c) AS min_c, MAX(c) AS max_c,
            MIN(d) AS min_d, MAX(d) AS max_d,
            MIN(e) AS min_e, MAX(e) AS max_e,
            MIN(a+e) AS mix_1, MIN(a)+MIN(e) AS mix_2
        FROM a GROUP BY a, b
        ORDER BY a NULLS FIRST, b NULLS FIRST""", a=a)


def test_sqlc_window_lead_lag_partition(ctx):
    a = rand_df(100, a=float, b=(int, 50), c=(str, 50))
    eq_sqlite(ctx, """
        SELECT
            LAG(b, 1) OVER (PARTITION BY c ORDER BY a) AS l1,
            LEAD(b, 1) OVER (PARTITION BY c ORDER BY a) AS l2,
            a, c FROM a
        ORDER BY a NULLS FIRST, c NULLS FIRST""", a=a)


def test_sqlc_window_ranks_partition(ctx):
    a = rand_df(100, a=int, b=(float, 50), c=(str, 50))
    eq_sqlite(ctx, """
        SELECT a, b,
            RANK() OVER (PARTITION BY a ORDER BY b NULLS LAST, c
                NULLS LAST) AS r,
            DENSE_RANK() OVER (PARTITION BY a ORDER BY b NULLS LAST, c
                NULLS LAST) AS d
        FROM a
        ORDER BY a, b NULLS LAST, c NULLS LAST""", a=a)


def test_sqlc_agg_count_group_expr(ctx):
    a = rand_df(100, a=int, b=(str, 50), c=(int, 30), d=(str, 40),
                e=(float, 40))
    eq_sqlite(ctx, """
        SELECT a, b, a+1 AS c, COUNT(c) AS cnt_c, COUNT(d) AS cnt_d
        FROM a GROUP BY a, b""", a=a)
    eq_sqlite(ctx, """
        SELECT b, COUNT(DISTINCT a) AS cnt_a, COUNT(e) AS cnt_e
        FROM a GROUP BY b""", a=a)


def test_sqlc_order_by_no_limit(ctx):
    a = rand_df(100, a=(int, 20), b=(str, 30))
    eq_sqlite(ctx, "SELECT * FROM a ORDER BY a NULLS LAST, b NULLS LAST",
              a=a)
    eq_sqlite(ctx, "SELECT b, a FROM a ORDER BY b DESC NULLS LAST, "
                   "a ASC NULLS FIRST", a=a)


def test_sqlc_join_multi(ctx):
    a = rand_df(100, a=(int, 40), b=(str, 40), c=(float, 40))
    b = rand_df(80, d=(float, 10), a=(int, 10), b=(str, 10))
    c = rand_df(80, dd=(float, 10), a=(int, 10), b=(str, 10))
    eq_sqlite(ctx, """
        SELECT a.*,d,dd FROM a
            INNER JOIN b ON a.a=b.a AND a.b=b.b
            INNER JOIN c ON a.a=c.a AND c.b=b.b
        ORDER BY a.a NULLS FIRST, a.b NULLS FIRST, a.c NULLS FIRST,
            dd NULLS FIRST, d NULLS FIRST""", a=a, b=b, c=c)


def test_sqlc_nested_window_filter(ctx):
    a = rand_df(100, a=float, b=(int, 50), c=(str, 50))
    eq_sqlite(ctx, """
        SELECT * FROM (
        SELECT *,
            ROW_NUMBER() OVER (PARTITION BY c ORDER BY b NULLS FIRST,
                a ASC NULLS LAST) AS r
        FROM a)
        WHERE r=1
        ORDER BY a NULLS LAST, b NULLS LAST, c NULLS LAST""", a=a)


def test_sqlc_union_three(ctx):
    a = rand_df(30, b=(int, 10), c=(str, 10))
    b = rand_df(80, b=(int, 50), c=(str, 50))
    c = rand_df(100, b=(int, 50), c=(str, 50))
    eq_sqlite(ctx, """
        SELECT * FROM a UNION SELECT * FROM b UNION SELECT * FROM c
        ORDER BY b NULLS FIRST, c NULLS FIRST""", a=a, b=b, c=c)
    eq_sqlite(ctx, """
        SELECT * FROM a UNION ALL SELECT * FROM b
            UNION ALL SELECT * FROM c
        ORDER BY b NULLS FIRST, c NULLS FIRST""", a=a, b=b, c=c)


def test_sqlc_window_sum_avg_irregular(ctx):
    a = rand_df(100, a=float, b=(int, 50), c=(str, 50))
    eq_sqlite(ctx, """
        SELECT a,b,
            SUM(b) OVER (ORDER BY a DESC
                ROWS BETWEEN 2 PRECEDING AND 1 PRECEDING) AS s6,
            AVG(b) OVER (ORDER BY a DESC
                ROWS BETWEEN 2 PRECEDING AND 1 FOLLOWING) AS a7,
            SUM(b) OVER (PARTITION BY c ORDER BY a
                ROWS BETWEEN 2 PRECEDING AND UNBOUNDED FOLLOWING) AS s8
        FROM a
        ORDER BY a NULLS FIRST, b NULLS FIRST, c NULLS FIRST""", a=a)


def test_sqlc_agg_count_full(ctx):
    a = rand_df(100, a=(int, 50), b=(str, 50), c=(int, 30), d=(str, 40),
                e=(float, 40))
    eq_sqlite(ctx, """
        SELECT a, b, a+1 AS c,
            COUNT(c) AS c_c, COUNT(DISTINCT c) AS cd_c,
            COUNT(d) AS c_d, COUNT(DISTINCT d) AS cd_d,
            COUNT(e) AS c_e, COUNT(DISTINCT a) AS cd_e
        FROM a GROUP BY a, b
        ORDER BY a NULLS FIRST, b NULLS FIRST""", a=a)
