"""Measure dsx call round-trip costs on the GPU box: (a) async op issue
cost, (b) count-returning op sync cost, (c) a full Q3 step phase split
with per-phase walltimes (what the 1.5 ms of non-kernel time is made of).
Run via gpurun."""
import sys, time
sys.path.insert(0, "/root/repo")
import numpy as np
from dask_sql_amd.runtime import Runtime
import dask_sql_amd.runtime as rt
from dask_sql_amd.physical.rex import OP_COL, OP_LIT_F64, OP_LT_F64

r = Runtime()
n = 1000
col = r.upload_column(np.random.rand(n))
prog = r.make_prog([(OP_COL, 0, 0), (OP_LIT_F64, 0, 0.5), (OP_LT_F64, 0, 0)])

# (a) async eval issue cost (no sync inside)
t0 = time.perf_counter()
outs = [r.eval(prog, [col], n, rt.BOOL8, with_validity=False)
        for _ in range(200)]
t1 = time.perf_counter()
r.synchronize()
print(f"async eval issue: {(t1-t0)/200*1e6:.1f} us/call")

# (b) filter (count sync) round trip
t0 = time.perf_counter()
for _ in range(200):
    p, cnt = r.filter(prog, [col], n)
    r.wrap_sel(p, cnt)
t1 = time.perf_counter()
print(f"filter sync call: {(t1-t0)/200*1e6:.1f} us/call")

# (c) Q3 step phase split
import pandas as pd
from dask_sql_amd.context import Context
from datagen import gen_q3, register_q3_tables, Q3_SQL
cust, orders, li = gen_q3()
c = Context()
register_q3_tables(c, cust, orders, li, persist=True)
for _ in range(3):
    c.sql(Q3_SQL).compute()

# phase split: filters / joins / agg / topk via targeted sub-queries
import contextlib
def timeit(f, k=10):
    f(); r2 = c._get_runtime(); r2.synchronize()
    t0 = time.perf_counter()
    for _ in range(k):
        f()
    r2.synchronize()
    return (time.perf_counter() - t0) / k * 1000

full = timeit(lambda: c.sql(Q3_SQL).compute())
nolimit = timeit(lambda: c.sql(Q3_SQL.split("ORDER BY")[0]).dc)
filters_only = timeit(lambda: (
    c.sql("SELECT c_custkey FROM customer WHERE c_mktsegment = 'BUILDING'").dc,
    c.sql("SELECT o_orderkey, o_custkey, o_orderdate, o_shippriority FROM orders WHERE o_orderdate < DATE '1995-03-15'").dc,
    c.sql("SELECT l_orderkey, l_extendedprice, l_discount FROM lineitem WHERE l_shipdate > DATE '1995-03-15'").dc))
plan_only = timeit(lambda: c._get_ral(Q3_SQL))
print(f"full step      : {full:.3f} ms")
print(f"no topk        : {nolimit:.3f} ms (topk+materialize = {full-nolimit:.3f})")
print(f"filters only   : {filters_only:.3f} ms")
print(f"plan cache hit : {plan_only*1000:.1f} us")
