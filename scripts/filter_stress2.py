import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from dask_sql_amd.runtime import Runtime

R = Runtime(0)
rng = np.random.default_rng(5)

# replicate the q3 scan shapes: customer i8 filter, orders i32, lineitem i32
cust_seg = rng.integers(0, 5, 1_500_000).astype(np.int8)
cust_key = np.arange(1_500_000, dtype=np.int64)
ord_date = rng.integers(8036, 10561, 15_000_000).astype(np.int32)
li_date = rng.integers(8036, 10561, 60_000_000).astype(np.int32)

c_seg = R.upload_column(cust_seg)
c_key = R.upload_column(cust_key)
o_date = R.upload_column(ord_date)
l_date = R.upload_column(li_date)

p_cust = R.make_prog([(1, 0, 0), (3, 0, 0), (34, 0, 0)])   # seg == 0
p_ord = R.make_prog([(1, 0, 0), (3, 0, 9204), (30, 0, 0)])  # date < 9204
p_li = R.make_prog([(1, 0, 0), (3, 0, 9204), (32, 0, 0)])   # date > 9204

exp_c = np.nonzero(cust_seg == 0)[0]
exp_o = np.nonzero(ord_date < 9204)[0]
exp_l = np.nonzero(li_date > 9204)[0]

def check(tag, cols, prog, n, exp):
    sel_ptr, count = R.filter(prog, cols, n)
    sel = R.wrap_sel(sel_ptr, count)
    ids = np.empty(count, dtype=np.uint32)
    R._download(sel.data, ids)
    ok = count == len(exp) and (ids < n).all() \
        and (ids.astype(np.int64) == exp).all()
    if not ok:
        mism = count - len(exp)
        inr = int((ids >= n).sum())
        print(f"{tag}: BAD count={count} exp={len(exp)} delta={mism} "
              f"oob={inr}")
        if count == len(exp):
            w = np.nonzero(ids.astype(np.int64) != exp)[0]
            print("  first mismatch positions:", w[:4], "ids:", ids[w[:4]],
                  "exp:", exp[w[:4]])
    # gather to mimic pipeline (allocates pool buffers incl. validity)
    g = R.gather(c_key if tag == "cust" else cols[0], sel.data, count,
                 force_validity=True)
    del g, sel
    return ok

bad = 0
for it in range(6):
    for tag, cols, prog, n, exp in [
        ("cust", [c_seg], p_cust, 1_500_000, exp_c),
        ("ord", [o_date], p_ord, 15_000_000, exp_o),
        ("li", [l_date], p_li, 60_000_000, exp_l),
    ]:
        if not check(tag, cols, prog, n, exp):
            bad += 1
print("bad:", bad)
