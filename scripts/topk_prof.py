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

# instrument _device_topk_impl stages by monkeypatching helpers it calls
import collections
acc = collections.Counter(); cnt = collections.Counter()
def wrap(mod, name):
    orig = getattr(mod, name)
    def f(*a, **k):
        t0 = time.perf_counter()
        r = orig(*a, **k)
        acc[name] += time.perf_counter() - t0; cnt[name] += 1
        return r
    setattr(mod, name, f)
    return orig
import dask_sql_amd.materialize as mat
wrap(rp, "_gather_table")
wrap(mat, "to_pandas")
orig_topk = rp._device_topk_impl
def timed_topk(*a, **k):
    t0 = time.perf_counter(); r = orig_topk(*a, **k)
    acc["_device_topk_impl"] += time.perf_counter() - t0
    cnt["_device_topk_impl"] += 1
    if r is not None:
        acc["_cand_rows"] += len(r.index) if hasattr(r, "index") else 0
    return r
rp._device_topk_impl = timed_topk
# also runtime pieces
from dask_sql_amd.runtime import Runtime
for m in ("filter", "gather", "_download"):
    orig = getattr(Runtime, m)
    def mk(orig, m):
        def f(self, *a, **k):
            t0 = time.perf_counter(); r = orig(self, *a, **k)
            acc["rt."+m] += time.perf_counter() - t0; cnt["rt."+m] += 1
            return r
        return f
    setattr(Runtime, m, mk(orig, m))

N = 10
t0 = time.perf_counter()
for _ in range(N):
    acc.clear(); cnt.clear()
    c.sql(Q3_SQL).compute()
wall = (time.perf_counter() - t0) / N
print(f"wall {wall*1000:.3f} ms; last-step pieces:")
for k_, v in acc.most_common():
    print(f"  {k_:22s} {v*1000:8.3f} ms ({cnt[k_]} calls)")
# how many candidate rows did the filter pass keep?
