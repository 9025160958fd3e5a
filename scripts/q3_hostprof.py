import sys, time, collections
sys.path.insert(0, "/root/repo")
from dask_sql_amd.runtime import Runtime
import dask_sql_amd.runtime as rtmod

acc = collections.Counter()
cnt = collections.Counter()


def wrap(name):
    orig = getattr(Runtime, name)

    def f(self, *a, **k):
        t0 = time.perf_counter()
        r = orig(self, *a, **k)
        acc[name] += time.perf_counter() - t0
        cnt[name] += 1
        return r

    setattr(Runtime, name, f)


for m in ("filter", "gather", "eval", "hash_build", "hash_probe",
          "hash_groupby", "minmax_i64", "keypack", "_download", "_upload_raw",
          "upload_column", "synchronize", "_malloc", "_free"):
    wrap(m)

from dask_sql_amd.context import Context
from datagen import gen_q3, register_q3_tables
from bench import WORKLOADS

cust, orders, li = gen_q3()
c = Context()
register_q3_tables(c, cust, orders, li, persist=True)
Q = WORKLOADS["q3_sf10"]["sql"]
for _ in range(2):
    c.sql(Q).compute()
acc.clear(); cnt.clear()
t0 = time.perf_counter()
N = 6
for _ in range(N):
    c.sql(Q).compute()
wall = (time.perf_counter() - t0) / N
print(f"wall/step: {wall*1000:.2f} ms")
tot = 0
for name, t in acc.most_common():
    print(f"  {name:14s} {t/N*1000:8.3f} ms/step  ({cnt[name]//N} calls)")
    tot += t / N
print(f"  accounted: {tot*1000:.2f} ms")


# python-level profile of one step (where the non-runtime host ms goes)
import cProfile, pstats, io
pr = cProfile.Profile()
pr.enable()
for _ in range(3):
    c.sql(Q).compute()
pr.disable()
sio = io.StringIO()
ps = pstats.Stats(pr, stream=sio).sort_stats("cumulative")
ps.print_stats(30)
print(sio.getvalue())
