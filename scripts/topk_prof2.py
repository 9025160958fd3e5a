import sys, time
sys.path.insert(0, "/root/repo")
import numpy as np, pandas as pd
from dask_sql_amd.context import Context
from datagen import gen_q3, register_q3_tables, Q3_SQL
import dask_sql_amd.physical.rel_plugins as rp

cust, orders, li = gen_q3()
c = Context()
register_q3_tables(c, cust, orders, li, persist=True)
for _ in range(3):
    c.sql(Q3_SQL).compute()

# time INSIDE the topk impl by sections using a modified copy
import dask_sql_amd.runtime as rtmod
T = {}
orig = rp._device_topk_impl
def probe(context, inp, below, keys, k):
    r = context._get_runtime()
    t = {}
    t0 = time.perf_counter()
    cc = inp.column_container
    n = inp.table.num_rows
    idx0, asc0, _ = keys[0]
    col0 = inp.table.col(cc.get_backend_by_frontend_name(cc.columns[idx0]))
    S = int(min(16384, n))
    cache = getattr(r, "_topk_sample_cache")
    sel = cache.get((n, S))
    sv, _v = r.gather(col0, sel.data, S).to_numpy()
    t["sample+dl"] = time.perf_counter() - t0; t0 = time.perf_counter()
    key_s = sv if asc0 else -sv
    rr = min(S - 1, max(int(np.ceil(k * S / n * 4)) + 8, k))
    thr_key = np.partition(key_s, rr)[rr]
    t["partition"] = time.perf_counter() - t0; t0 = time.perf_counter()
    out = orig(context, inp, below, keys, k)
    t["rest(full impl)"] = time.perf_counter() - t0
    for k_, v in t.items():
        T[k_] = T.get(k_, 0) + v
    T["n_cand?"] = n
    return out
rp._device_topk_impl = probe

N = 10
for _ in range(N):
    T.clear()
    c.sql(Q3_SQL).compute()
for k_, v in T.items():
    print(f"  {k_:18s} {v*1000 if k_ != 'n_cand?' else v:10.3f}")

# and: how big is the candidate set + time the filter alone
from dask_sql_amd.physical.rex import OP_COL, OP_LIT_F64, OP_GE_F64
res = c.sql(Q3_SQL.split("ORDER BY")[0])
dc = res.dc
cc = dc.column_container
col0 = dc.table.col(cc.get_backend_by_frontend_name(cc.columns[1]))
n = dc.table.num_rows
print("groups:", n)
r = c._get_runtime()
prog = r.make_prog([(OP_COL, 0, 0), (OP_LIT_F64, 0, 400000.0), (OP_GE_F64, 0, 0)])
t0 = time.perf_counter()
for _ in range(20):
    p, cnt2 = r.filter(prog, [col0], n)
    r.wrap_sel(p, cnt2)
print(f"threshold filter over {n}: {(time.perf_counter()-t0)/20*1000:.3f} ms, cnt={cnt2}")
