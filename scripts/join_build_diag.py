import sys, time
sys.path.insert(0, "/root/repo")
import numpy as np
from dask_sql_amd.runtime import Runtime

rt = Runtime(0)
rt.prof_enable(True)
for n, space in [(300_000, 1_500_000), (3_750_000, 15_000_000),
                 (3_750_000, 3_750_000), (10_000_000, 10_000_000)]:
    rng = np.random.default_rng(1)
    codes = rng.choice(space, size=n, replace=False).astype(np.int64) \
        if space >= n else rng.integers(0, space, n).astype(np.int64)
    col = rt.upload_column(codes)
    rt.prof_reset()
    for _ in range(3):
        t = rt.hash_build(col)
        rt.hash_table_free(t)
    rt.synchronize()
    print(n, space, rt.prof_get())
