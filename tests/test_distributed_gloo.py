"""world_size-2 CPU (gloo) coverage of the multi-GPU shuffle orchestration
(dask_sql_amd/distributed.py — SURVEY §8e): bucket exchange plumbing and the
partial-merge algebra. The GPU kernels themselves are covered by -m gpu;
these tests pin the communication layer the driver's 8-GPU run exercises."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from tests.conftest import REPO


def _run_exchange(rank, world, port, results):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dask_sql_amd.distributed import exchange_buckets

        # rank r sends [r*100 + b*10 .. ) to bucket/rank b, 3+b+r rows each
        splits = [3 + b + rank for b in range(world)]
        vals = []
        for b in range(world):
            vals += [rank * 100 + b * 10 + i for i in range(splits[b])]
        t = torch.tensor(vals, dtype=torch.int64)
        received, out_splits = exchange_buckets([t], splits)
        got = received[0].tolist()
        # expected: from each rank s, its bucket `rank` contents
        exp = []
        for s in range(world):
            cnt = 3 + rank + s
            exp += [s * 100 + rank * 10 + i for i in range(cnt)]
        assert got == exp, (rank, got, exp)
        assert out_splits == [3 + rank + s for s in range(world)]
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def _run_partial_merge(rank, world, port, results):
    """CPU mirror of merge_groupby_partials' algebra: partials exchanged by
    key hash → each rank owns a disjoint key set → local re-aggregate;
    union across ranks equals the global aggregation."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dask_sql_amd.distributed import bucket_of_np, exchange_buckets

        rng = np.random.default_rng(100 + rank)
        keys = rng.integers(0, 50, 1000).astype(np.int64)
        vals = rng.random(1000)
        # local partial aggregation (what the local fused kernel produces)
        pk = np.unique(keys)
        psum = np.array([vals[keys == k].sum() for k in pk])
        pcnt = np.array([(keys == k).sum() for k in pk], dtype=np.int64)
        # bucket partials by key hash (mirrors dsx_partition's function)
        b = bucket_of_np(pk.astype(np.uint64), world)
        order = np.argsort(b, kind="stable")
        splits = [int((b == r).sum()) for r in range(world)]
        tk = torch.tensor(pk[order])
        ts = torch.tensor(psum[order])
        tc = torch.tensor(pcnt[order])
        (rk, rs, rc), _ = exchange_buckets([tk, ts, tc], splits)

        rk = rk.numpy()
        # every received key must hash to MY bucket (disjoint ownership)
        assert (bucket_of_np(rk.astype(np.uint64), world) == rank).all()
        # local re-aggregation
        uk = np.unique(rk)
        merged = {int(k): (rs.numpy()[rk == k].sum(),
                           int(rc.numpy()[rk == k].sum())) for k in uk}
        # gather to rank 0 and compare against the global truth
        all_m = [None] * world
        dist.all_gather_object(all_m, merged)
        if rank == 0:
            combined = {}
            for m in all_m:
                for k, v in m.items():
                    assert k not in combined, "key owned by two ranks"
                    combined[k] = v
            # global truth: regenerate every rank's rows
            gk = []
            gv = []
            for r in range(world):
                rngr = np.random.default_rng(100 + r)
                kr = rngr.integers(0, 50, 1000).astype(np.int64)
                vr = rngr.random(1000)
                gk.append(kr)
                gv.append(vr)
            gk = np.concatenate(gk)
            gv = np.concatenate(gv)
            for k in np.unique(gk):
                s, c = combined[int(k)]
                assert c == (gk == k).sum()
                assert abs(s - gv[gk == k].sum()) < 1e-9
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def _spawn(fn, world=2):
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=fn, args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(120)
        assert dict(results) == {r: "ok" for r in range(world)}, \
            dict(results)


def test_exchange_buckets_gloo():
    _spawn(_run_exchange)


def test_exchange_buckets_gloo_world3():
    # odd world size: uneven splits, non-power-of-two bucket hash
    _spawn(_run_exchange, world=3)


def test_partial_merge_algebra_gloo():
    _spawn(_run_partial_merge)


def test_partial_merge_algebra_gloo_world3():
    _spawn(_run_partial_merge, world=3)


def test_q1_merge_partials_cpu():
    """q1_merge_partials (incl. the count-weighted AVG recombination —
    VERDICT r1 weak#1a) over row slices equals the oracle on the full
    table. Pure host logic, no process group needed."""
    import sys
    sys.path.insert(0, str(REPO))
    from datagen import gen_lineitem_q1
    from dask_sql_amd.distributed import q1_merge_partials
    from oracle.tpch import oracle_q1
    import pandas as pd

    li = gen_lineitem_q1(n=200_000, seed=7)
    world = 4
    partials = [oracle_q1(li.iloc[r::world].reset_index(drop=True))
                for r in range(world)]
    merged = q1_merge_partials(partials)
    merged = merged.sort_values(["l_returnflag", "l_linestatus"]
                                ).reset_index(drop=True)
    exp = oracle_q1(li)
    assert merged["count_order"].tolist() == exp["count_order"].tolist()
    for col in ["sum_qty", "sum_base_price", "sum_disc_price", "sum_charge",
                "avg_qty", "avg_price", "avg_disc"]:
        got = merged[col].to_numpy(dtype=np.float64)
        want = exp[col].to_numpy(dtype=np.float64)
        assert np.allclose(got, want, rtol=1e-9), col


def _run_q1_merge(rank, world, port, results):
    """World-2 end-to-end Q1 merge: per-rank oracle partials gathered and
    merged must equal the single-process oracle frame."""
    import sys
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        sys.path.insert(0, str(REPO))
        from datagen import gen_lineitem_q1
        from dask_sql_amd.distributed import q1_merge_partials
        from oracle.tpch import oracle_q1
        import pandas as pd

        n_total = 120_000
        full = gen_lineitem_q1(n=n_total, seed=11)
        mine = full.iloc[rank::world].reset_index(drop=True)
        part = oracle_q1(mine)
        gathered = [None] * world
        dist.all_gather_object(gathered, part)
        if rank == 0:
            merged = q1_merge_partials(gathered).sort_values(
                ["l_returnflag", "l_linestatus"]).reset_index(drop=True)
            exp = oracle_q1(full)
            assert merged["count_order"].tolist() == \
                exp["count_order"].tolist()
            for col in ["sum_qty", "sum_disc_price", "sum_charge",
                        "avg_qty", "avg_price", "avg_disc"]:
                assert np.allclose(merged[col].to_numpy(np.float64),
                                   exp[col].to_numpy(np.float64),
                                   rtol=1e-9), col
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def _run_q3_pipeline(rank, world, port, results):
    """World-2 mirror of distributed.q3_distributed with oracle compute:
    slice → filter → exchange by custkey → join → exchange by orderkey →
    join+groupby → per-rank top-10 → global merge; the merged frame must
    equal the single-process oracle Q3."""
    import sys
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        sys.path.insert(0, str(REPO))
        from datagen import gen_q3
        from dask_sql_amd.distributed import bucket_of_np, exchange_buckets
        from oracle.frame import oracle_filter, oracle_groupby, oracle_join
        from oracle.tpch import D_1995_03_15, oracle_q3
        import pandas as pd

        cust, orders, li = gen_q3(sf_rows=(6_000, 30_000, 120_000))
        mc = cust.iloc[rank::world].reset_index(drop=True)
        mo = orders.iloc[rank::world].reset_index(drop=True)
        ml = li.iloc[rank::world].reset_index(drop=True)

        def exch(df, keycol):
            b = bucket_of_np(df[keycol].to_numpy(np.uint64), world)
            order = np.argsort(b, kind="stable")
            splits = [int((b == r).sum()) for r in range(world)]
            ts = [torch.tensor(df[c].to_numpy()[order]) for c in df.columns]
            recv, _ = exchange_buckets(ts, splits)
            return pd.DataFrame({c: t.numpy() for c, t in
                                 zip(df.columns, recv)})

        c_f = oracle_filter(mc, mc["c_mktsegment"] == 0)[["c_custkey"]]
        o_f = oracle_filter(mo, mo["o_orderdate"] < D_1995_03_15)[
            ["o_orderkey", "o_custkey", "o_orderdate", "o_shippriority"]]
        l_f = oracle_filter(ml, ml["l_shipdate"] > D_1995_03_15)[
            ["l_orderkey", "l_extendedprice", "l_discount"]]
        c_x = exch(c_f, "c_custkey")
        o_x = exch(o_f, "o_custkey")
        co = oracle_join(o_x, c_x, [1], [0], "INNER")
        co.columns = ["o_orderkey", "o_custkey", "o_orderdate",
                      "o_shippriority", "c_custkey"]
        co = co[["o_orderkey", "o_orderdate", "o_shippriority"]]
        co_x = exch(co, "o_orderkey")
        l_x = exch(l_f, "l_orderkey")
        col = oracle_join(l_x, co_x, [0], [0], "INNER")
        col.columns = ["l_orderkey", "l_extendedprice", "l_discount",
                       "o_orderkey", "o_orderdate", "o_shippriority"]
        col = col.assign(
            revenue=col["l_extendedprice"] * (1 - col["l_discount"]))
        out = oracle_groupby(
            col, ["l_orderkey", "o_orderdate", "o_shippriority"],
            [("revenue", "revenue", "sum", None, False)])
        out = out[["l_orderkey", "revenue", "o_orderdate", "o_shippriority"]]
        top = out.sort_values(["revenue", "o_orderdate"],
                              ascending=[False, True],
                              kind="mergesort").head(10)
        tops = [None] * world
        dist.all_gather_object(tops, top)
        if rank == 0:
            allt = pd.concat(tops).sort_values(
                ["revenue", "o_orderdate"], ascending=[False, True],
                kind="mergesort").head(10).reset_index(drop=True)
            exp = oracle_q3(cust, orders, li)
            assert allt["l_orderkey"].tolist() == exp["l_orderkey"].tolist()
            assert np.allclose(allt["revenue"].to_numpy(np.float64),
                               exp["revenue"].to_numpy(np.float64),
                               rtol=1e-9)
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def test_q1_merge_gloo():
    _spawn(_run_q1_merge)


def test_q3_pipeline_merge_gloo():
    _spawn(_run_q3_pipeline)


def _run_broadcast_exchange(rank, world, port, results):
    """Broadcast-join build-side replication (sql.join.broadcast →
    allgather_device_columns' exchange pattern): every rank must end up
    with the concatenation of all ranks' rows."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dask_sql_amd.distributed import exchange_buckets

        n = 5 + rank
        t = torch.arange(n, dtype=torch.int64) + rank * 100
        received, out_splits = exchange_buckets([t.repeat(world)],
                                                [n] * world)
        exp = []
        for s in range(world):
            exp += [s * 100 + i for i in range(5 + s)]
        assert received[0].tolist() == exp, (rank, received[0].tolist())
        assert out_splits == [5 + s for s in range(world)]
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()


def test_broadcast_exchange_gloo():
    _spawn(_run_broadcast_exchange)


def test_broadcast_exchange_gloo_world3():
    _spawn(_run_broadcast_exchange, world=3)


def test_q1_merge_partials_edges():
    """q1_merge_partials: None entries (non-zero ranks), single partial
    identity, and empty-group alignment."""
    import sys
    sys.path.insert(0, str(REPO))
    from datagen import gen_lineitem_q1
    from dask_sql_amd.distributed import q1_merge_partials
    from oracle.tpch import oracle_q1

    li = gen_lineitem_q1(n=50_000, seed=13)
    full = oracle_q1(li)
    # identity: merging one partial (+ Nones) reproduces it
    m = q1_merge_partials([full, None, None])
    m = m.sort_values(["l_returnflag", "l_linestatus"]).reset_index(drop=True)
    assert m["count_order"].tolist() == full["count_order"].tolist()
    np.testing.assert_allclose(m["avg_disc"], full["avg_disc"], rtol=1e-12)
    # a rank whose slice lacks some groups entirely
    p1 = oracle_q1(li.iloc[:1000].reset_index(drop=True))
    p2 = oracle_q1(li.iloc[1000:].reset_index(drop=True))
    m2 = q1_merge_partials([p1, p2]).sort_values(
        ["l_returnflag", "l_linestatus"]).reset_index(drop=True)
    assert m2["count_order"].tolist() == full["count_order"].tolist()
    np.testing.assert_allclose(m2["sum_charge"], full["sum_charge"],
                               rtol=1e-9)
    np.testing.assert_allclose(m2["avg_qty"], full["avg_qty"], rtol=1e-9)
