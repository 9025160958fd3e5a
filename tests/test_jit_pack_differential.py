"""CPU differential test of the JIT key-pack codegen (jit.inc
jit_emit_pack) against a direct restatement of the pack_key spec
(csrc/dsxhip.hip:1248 + dsx_keypack stride construction): random key
configurations are emitted via dsx_jit_pack_source, gcc-compiled behind
host shims, and run over random columns with NULLs. The r2 radix-join
nullable-pack bug was exactly a jit-vs-static disagreement of this kind —
this pins the pair without a GPU."""
import ctypes as ct
import struct
import subprocess
import tempfile
from pathlib import Path

import numpy as np
import pytest

from dask_sql_amd import runtime as rt

PRELUDE = r"""
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
typedef int64_t i64;
typedef uint64_t u64;
typedef unsigned char u8;
#define __device__
#define __forceinline__ inline
struct ColsArg {
  const void* data[16];
  const u8* validity[16];
  int dtype[16];
  int ncols;
};
static inline i64 __double_as_longlong(double d) {
  i64 x; memcpy(&x, &d, 8); return x;
}
"""


def spec_pack(cols, keys, row):
    """Direct restatement of pack_key (dsxhip.hip:1248) + the stride rule
    of dsx_keypack (dsxhip.hip:1301-1313)."""
    code = np.uint64(0)
    stride = np.uint64(1)
    for (ci, mn, rng, nullable, mode) in keys:
        vals, valid = cols[ci]
        ok = True if valid is None else bool(valid[row])
        if mode == 1:
            x = float(vals[row])
            if not ok or x != x:
                part = np.uint64(0)
            else:
                if x == 0.0:
                    x = 0.0
                bits = struct.unpack("<q", struct.pack("<d", x))[0]
                part = np.uint64(bits & ((1 << 64) - 1)) + np.uint64(1)
            code = code + part * stride
            continue
        v = int(vals[row])
        if nullable:
            part = np.uint64((v - mn) + 1) if ok else np.uint64(0)
        else:
            part = np.uint64((v - mn) & ((1 << 64) - 1))
        code = code + part * stride
        stride = stride * np.uint64(rng + (1 if nullable else 0))
    return int(code)


@pytest.fixture(scope="module")
def packs():
    lib = rt._load_lib()
    lib.dsx_jit_pack_source.argtypes = [
        ct.POINTER(rt._KeySpec), ct.c_int, ct.POINTER(ct.c_int32),
        ct.POINTER(ct.c_uint8), ct.c_int, ct.c_char_p, ct.c_int64]
    lib.dsx_jit_pack_source.restype = ct.c_int

    rng = np.random.default_rng(777)
    n_rows = 64
    # columns: 0 i64 dense, 1 i64 nullable, 2 i32 nullable, 3 i8 dense,
    # 4 f64 nullable (for the f64-bits single-key mode)
    iv0 = rng.integers(-100, 100, n_rows).astype(np.int64)
    iv1 = rng.integers(0, 50, n_rows).astype(np.int64)
    v1 = (rng.random(n_rows) > 0.3).astype(np.uint8)
    iv2 = rng.integers(-8, 8, n_rows).astype(np.int32)
    v2 = (rng.random(n_rows) > 0.3).astype(np.uint8)
    iv3 = rng.integers(0, 4, n_rows).astype(np.int8)
    fv4 = np.round(rng.uniform(-5, 5, n_rows), 2)
    fv4[rng.random(n_rows) < 0.2] = np.nan
    fv4[0] = 0.0
    fv4[1] = -0.0  # canonical-zero case
    v4 = (rng.random(n_rows) > 0.2).astype(np.uint8)
    cols = [(iv0, None), (iv1, v1), (iv2, v2), (iv3, None), (fv4, v4)]
    dtypes = [rt.I64, rt.I64, rt.I32, rt.I8, rt.F64]
    hasv = [0, 1, 1, 0, 1]

    configs = []
    # multi-key integer configs
    for _ in range(12):
        nk = int(rng.integers(1, 4))
        ks = []
        pool = [0, 1, 2, 3]
        rng.shuffle(pool)
        for j in range(nk):
            ci = pool[j]
            lo = {0: -100, 1: 0, 2: -8, 3: 0}[ci]
            hi = {0: 100, 1: 50, 2: 8, 3: 4}[ci]
            nullable = 1 if hasv[ci] and rng.random() < 0.7 else 0
            ks.append((ci, lo, hi - lo, nullable, 0))
        configs.append(ks)
    # single f64-bits key
    configs.append([(4, 0, 0, 1, 1)])

    sources = []
    kept = []
    for ks in configs:
        arr = (rt._KeySpec * len(ks))()
        for j, (ci, mn, rg, nu, mo) in enumerate(ks):
            arr[j].col, arr[j].min, arr[j].range = ci, mn, rg
            arr[j].nullable, arr[j].mode = nu, mo
        dt = (ct.c_int32 * 5)(*dtypes)
        hv = (ct.c_uint8 * 5)(*hasv)
        buf = ct.create_string_buffer(1 << 14)
        rc = lib.dsx_jit_pack_source(arr, len(ks), dt, hv, 5, buf,
                                     len(buf))
        assert rc == 0, rc
        i = len(kept)
        sources.append(buf.value.decode().replace("jit_pack",
                                                  f"jit_pack_{i}"))
        kept.append(ks)

    td = Path(tempfile.mkdtemp(prefix="dsx_packdiff_"))
    calls = [f"    case {i}: printf(\"%llu\\n\", (unsigned long long)"
             f"jit_pack_{i}(C, r)); break;" for i in range(len(kept))]
    src = PRELUDE + "\n".join(sources) + f"""
int main(int argc, char** argv) {{
  int idx = atoi(argv[1]);
  static i64 c0[{n_rows}], c1[{n_rows}];
  static int c2[{n_rows}];
  static signed char c3[{n_rows}];
  static double c4[{n_rows}];
  static u8 v1[{n_rows}], v2[{n_rows}], v4[{n_rows}];
  FILE* f = fopen(argv[2], "rb");
  fread(c0, 8, {n_rows}, f); fread(c1, 8, {n_rows}, f);
  fread(c2, 4, {n_rows}, f); fread(c3, 1, {n_rows}, f);
  fread(c4, 8, {n_rows}, f);
  fread(v1, 1, {n_rows}, f); fread(v2, 1, {n_rows}, f);
  fread(v4, 1, {n_rows}, f);
  fclose(f);
  struct ColsArg C;
  C.ncols = 5;
  C.data[0]=c0; C.data[1]=c1; C.data[2]=c2; C.data[3]=c3; C.data[4]=c4;
  C.validity[0]=0; C.validity[1]=v1; C.validity[2]=v2; C.validity[3]=0;
  C.validity[4]=v4;
  for (i64 r = 0; r < {n_rows}; r++) {{
    switch (idx) {{
{chr(10).join(calls)}
    }}
  }}
  return 0;
}}
"""
    cpath = td / "packs.c"
    cpath.write_text(src)
    exe = td / "packs"
    r = subprocess.run(["gcc", "-O1", "-x", "c++", str(cpath), "-o",
                        str(exe), "-lstdc++"], capture_output=True,
                       text=True)
    assert r.returncode == 0, r.stderr[:4000]

    data = td / "cols.bin"
    with open(data, "wb") as f:
        f.write(iv0.astype("<i8").tobytes())
        f.write(iv1.astype("<i8").tobytes())
        f.write(iv2.astype("<i4").tobytes())
        f.write(iv3.astype("i1").tobytes())
        f.write(fv4.astype("<f8").tobytes())
        f.write(v1.tobytes())
        f.write(v2.tobytes())
        f.write(v4.tobytes())

    def run(idx):
        out = subprocess.run([str(exe), str(idx), str(data)],
                             capture_output=True, text=True, timeout=60)
        assert out.returncode == 0
        return [int(x) for x in out.stdout.split()]

    return kept, cols, run


def test_jit_pack_matches_spec(packs):
    kept, cols, run = packs
    for i, ks in enumerate(kept):
        got = run(i)
        for row, g in enumerate(got):
            want = spec_pack(cols, ks, row)
            assert g == want, (i, ks, row, g, want)


def test_order_mode_bits_refused():
    lib = rt._load_lib()
    arr = (rt._KeySpec * 1)()
    arr[0].col, arr[0].min, arr[0].range = 0, 0, 10
    arr[0].nullable, arr[0].mode = 0, 2  # DESC bit
    dt = (ct.c_int32 * 1)(rt.I64)
    hv = (ct.c_uint8 * 1)(0)
    buf = ct.create_string_buffer(1 << 14)
    rc = lib.dsx_jit_pack_source(arr, 1, dt, hv, 1, buf, len(buf))
    assert rc == 0
    assert b"#error" in buf.value  # loud hiprtc failure, not silent mispack
