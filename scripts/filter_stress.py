import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from dask_sql_amd.runtime import Runtime
from dask_sql_amd import runtime as rt

R = Runtime(0)
rng = np.random.default_rng(3)
n = 60_000_000
x = rng.integers(0, 2500, n).astype(np.int32) + 8000  # shipdate-like
col = R.upload_column(x)
exp_ids = np.nonzero(x > 9204)[0]
print("expect", len(exp_ids))
prog = R.make_prog([(1, 0, 0), (3, 0, 9204), (32, 0, 0)])  # col0 > 9204 (i64)
bad = 0
for i in range(12):
    sel_ptr, count = R.filter(prog, [col], n)
    sel = R.wrap_sel(sel_ptr, count)
    ids = np.empty(count, dtype=np.uint32)
    R._download(sel.data, ids)
    ok_count = count == len(exp_ids)
    in_range = (ids < n).all()
    exact = in_range and ok_count and (ids.astype(np.int64) == exp_ids).all()
    if not exact:
        bad += 1
        badidx = np.nonzero(ids.astype(np.int64) != exp_ids[:len(ids)])[0][:5] \
            if ok_count else []
        print(f"iter {i}: count={count} expected={len(exp_ids)} "
              f"in_range={in_range} first_bad={badidx}")
    del sel
print("bad iters:", bad)
