import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from dask_sql_amd.runtime import Runtime

rt = Runtime(0)
rt.prof_enable(True)

cases = {
    # (codes array, label)
    "random_unique_300k": np.random.default_rng(1).choice(
        1_500_000, 300_000, replace=False).astype(np.int64),
    "consecutive_300k": np.arange(300_000, dtype=np.int64),
    "strided_300k_of_1p5m": np.arange(0, 1_500_000, 5, dtype=np.int64),
    "orders_like_1p5m_of_15m": np.sort(np.random.default_rng(2).choice(
        15_000_000, 1_500_000, replace=False)).astype(np.int64),
    "random_10m": np.random.default_rng(3).permutation(
        10_000_000).astype(np.int64),
}
for name, codes in cases.items():
    col = rt.upload_column(codes)
    rt.prof_reset()
    for _ in range(3):
        t = rt.hash_build(col)
        rt.hash_table_free(t)
    rt.synchronize()
    p = rt.prof_get().get("k_hash_build", {})
    print(f"{name:28s} n={len(codes):9d} "
          f"{p.get('ms', 0)/max(p.get('launches',1),1):8.3f} ms/launch")
