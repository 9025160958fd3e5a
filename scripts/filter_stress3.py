import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from dask_sql_amd.runtime import Runtime
from dask_sql_amd import runtime as rt

R = Runtime(0)
rng = np.random.default_rng(5)
cust_seg = rng.integers(0, 5, 1_500_000).astype(np.int8)
c_seg = R.upload_column(cust_seg)
prog = R.make_prog([(1, 0, 0), (3, 0, 0), (34, 0, 0)])  # seg == 0

ev = R.eval(prog, [c_seg], len(cust_seg), rt.BOOL8, with_validity=False)
got = np.empty(len(cust_seg), dtype=np.uint8)
R._download(ev.data, got)
exp = (cust_seg == 0).astype(np.uint8)
mism = np.nonzero(got != exp)[0]
print("eval mismatches:", len(mism))
if len(mism):
    print("first idx:", mism[:8])
    print("host seg vals:", cust_seg[mism[:8]])
    print("gpu pred:", got[mism[:8]])
    # dump the raw device bytes at those positions
    raw = np.empty(len(cust_seg), dtype=np.int8)
    R._download(c_seg.data, raw)
    print("device col bytes:", raw[mism[:8]])
    print("upload equal:", (raw == cust_seg).all())
    bad = np.nonzero(raw != cust_seg)[0]
    print("upload bad count:", len(bad), bad[:5],
          "host:", cust_seg[bad[:5]] if len(bad) else "",
          "dev:", raw[bad[:5]] if len(bad) else "")
