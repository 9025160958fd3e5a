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
