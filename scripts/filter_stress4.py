import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from dask_sql_amd.runtime import Runtime

R = Runtime(0)
rng = np.random.default_rng(5)
cust_seg = rng.integers(0, 5, 1_500_000).astype(np.int8)
c_seg = R.upload_column(cust_seg)
prog = R.make_prog([(1, 0, 0), (3, 0, 0), (34, 0, 0)])  # seg == 0
exp = np.nonzero(cust_seg == 0)[0]

sel_ptr, count = R.filter(prog, [c_seg], len(cust_seg))
sel = R.wrap_sel(sel_ptr, count)
ids = np.empty(count, dtype=np.uint32)
R._download(sel.data, ids)
print("count", count, "exp", len(exp), "delta", count - len(exp))
print("unique", np.unique(ids).size)
ids64 = ids.astype(np.int64)
n_common = min(count, len(exp))
div = np.nonzero(ids64[:n_common] != exp[:n_common])[0]
print("first divergence at:", div[0] if len(div) else None)
if len(div):
    d = div[0]
    print("around divergence ids:", ids64[d-2:d+6])
    print("around divergence exp:", exp[d-2:d+6])
    # which rows are extra?
    extra = np.setdiff1d(ids64, exp)
    print("extra rows:", len(extra), extra[:10])
    print("extra row values:", cust_seg[extra[:10]])
    print("extra rows word-ids:", (extra[:10] // 64))
    missing = np.setdiff1d(exp, ids64)
    print("missing rows:", len(missing), missing[:10])
    # are extras duplicates of real ids?
    dup = count - np.unique(ids).size
    print("dup count:", dup)
