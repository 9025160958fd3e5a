"""Subprocess body for the fallback-knob parity test: runs a small
groupby + join + filter with the CURRENT env (e.g. DSX_DISABLE_JIT=1)
and prints checksums; the parent compares against the default path.
Torch-free (Context + ctypes only) to keep startup cheap."""
import json
import sys

sys.path.insert(0, sys.argv[1])
import numpy as np
import pandas as pd

from dask_sql_amd.context import Context

rng = np.random.default_rng(77)
n = 2_000_000
key = rng.integers(0, 50_000, n).astype(np.int64)
val = rng.random(n)
bk = rng.choice(100_000, 60_000, replace=False).astype(np.int64)
pk = rng.integers(0, 100_000, n).astype(np.int64)

c = Context()
c.create_table("t", pd.DataFrame({"key": key, "x": val}))
c.create_table("b", pd.DataFrame({"k": bk, "bv": bk * 2}))
c.create_table("p", pd.DataFrame({"k": pk}))

g = c.sql("SELECT key, SUM(x) AS s, COUNT(*) AS c FROM t "
          "WHERE x < 0.5 GROUP BY key").compute()
g = g.sort_values("key").reset_index(drop=True)
j = c.sql("SELECT p.k, b.bv FROM p JOIN b ON p.k = b.k").compute()
out = {
    "g_rows": int(len(g)),
    "g_count": int(g["c"].sum()),
    "g_sum": float(g["s"].sum()),
    "g_keysum": int(g["key"].astype(np.int64).sum()),
    "j_rows": int(len(j)),
    "j_ksum": int(j["k"].astype(np.int64).sum()),
    "j_bvsum": int(j["bv"].astype(np.int64).sum()),
}
print(json.dumps(out))
