"""Differential test of the PRODUCTION JIT codegen on CPU: for random
compiled expressions, ask libdsxhip.so for the exact C source its hipRTC
path generates (dsx_jit_expr_source — emission only, no HIP calls), compile
that source with gcc behind small host shims for the device intrinsics, run
it over real column data, and compare value+validity against the faithful
VM model (tests/vm_interp.py). This pins jit.inc's JitExprGen semantics
(Kleene AND/OR, guarded division, validity strings, conversions) without a
GPU; the remaining gap — hipRTC vs gcc floating-point codegen — is covered
by the gpu-marked parity suites."""
import ctypes as ct
import math
import subprocess
import tempfile
from pathlib import Path

import numpy as np
import pytest

from dask_sql_amd import runtime as rt
from dask_sql_amd.physical import rex as R
from tests.test_vm_differential import _dev_cols, _make_cols, gen
from tests.vm_interp import interp

HARNESS_PRELUDE = r"""
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <math.h>
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
static inline double __longlong_as_double(i64 x) {
  double d; memcpy(&d, &x, 8); return d;
}
static inline double __ocml_exp_f64(double x) { return exp(x); }
static inline double __ocml_log_f64(double x) { return log(x); }
static inline double __ocml_pow_f64(double a, double b) { return pow(a, b); }
static inline double __ocml_sin_f64(double x) { return sin(x); }
static inline double __ocml_cos_f64(double x) { return cos(x); }
static inline double __ocml_tan_f64(double x) { return tan(x); }
static inline double __ocml_asin_f64(double x) { return asin(x); }
static inline double __ocml_acos_f64(double x) { return acos(x); }
static inline double __ocml_atan_f64(double x) { return atan(x); }
static inline double __ocml_atan2_f64(double a, double b) {
  return atan2(a, b);
}
static inline i64 jit_absl(i64 v) { return v < 0 ? -v : v; }
/* AMD v_cvt f64->i64 saturates and maps NaN to 0; x86 cvttsd2si gives
   INT64_MIN — route every (i64) cast through this to match the device */
template <class T> static inline i64 dsx_cvt(T x) { return (i64)x; }
static inline i64 dsx_cvt(double x) {
  if (isnan(x)) return 0;
  if (x >= 9223372036854775807.0) return INT64_MAX;
  if (x <= -9223372036854775808.0) return INT64_MIN;
  return (i64)x;
}
static inline i64 dsx_cvt(float x) { return dsx_cvt((double)x); }
static inline void jit_civil(i64 days, int* y, int* m, int* d) {
  i64 z = days + 719468;
  i64 era = (z >= 0 ? z : z - 146096) / 146097;
  i64 doe = z - era * 146097;
  i64 yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  i64 yy = yoe + era * 400;
  i64 doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  i64 mp = (5 * doy + 2) / 153;
  *d = (int)(doy - (153 * mp + 2) / 5 + 1);
  *m = (int)(mp < 10 ? mp + 3 : mp - 9);
  *y = (int)(yy + (*m <= 2));
}
static inline i64 jit_year(i64 v) { int y,m,d; jit_civil(v,&y,&m,&d); return y; }
static inline i64 jit_month(i64 v) { int y,m,d; jit_civil(v,&y,&m,&d); return m; }
static inline i64 jit_day(i64 v) { int y,m,d; jit_civil(v,&y,&m,&d); return d; }
"""


def _devicify_casts(src):
    """Rewrite every `(i64)( ... )` cast to `dsx_cvt(( ... ))` with
    balanced parens so host gcc reproduces the device's saturating
    float->int conversion."""
    out = []
    i = 0
    pat = "(i64)("
    while True:
        j = src.find(pat, i)
        if j < 0:
            out.append(src[i:])
            return "".join(out)
        out.append(src[i:j])
        k = j + len(pat)
        depth = 1
        while depth:
            ch = src[k]
            if ch == "(":
                depth += 1
            elif ch == ")":
                depth -= 1
            k += 1
        inner = src[j + len(pat):k - 1]
        out.append("dsx_cvt((" + _devicify_casts(inner) + "))")
        i = k


def _emit_source(lib, prog_tuple):
    arr, n = rt.Runtime.make_prog(prog_tuple)
    dt = (ct.c_int32 * 5)(rt.I64, rt.I64, rt.F64, rt.F64, rt.I64)
    hv = (ct.c_uint8 * 5)(0, 1, 0, 1, 0)
    buf = ct.create_string_buffer(1 << 16)
    rc = lib.dsx_jit_expr_source(arr, n, dt, hv, 5, buf, len(buf))
    if rc < 0:
        return None, None
    return buf.value.decode(), ("d" if rc == 1 else "l")


@pytest.fixture(scope="module")
def harness():
    """Build one gcc binary holding every generated expression fn; returns
    a runner(idx, rows) -> [(value, valid)]."""
    lib = rt._load_lib()
    lib.dsx_jit_expr_source.argtypes = [
        ct.POINTER(rt._Instr), ct.c_int, ct.POINTER(ct.c_int32),
        ct.POINTER(ct.c_uint8), ct.c_int, ct.c_char_p, ct.c_int64]
    lib.dsx_jit_expr_source.restype = ct.c_int

    rng = np.random.default_rng(4321)
    n_rows = 40
    cols = _make_cols(rng, n_rows)
    # overwrite col 0 with day-scale / ns-scale values so the datetime
    # programs below see plausible calendar inputs (dense int64 col)
    cols[0] = (rng.integers(-30000, 60000, n_rows).astype(np.int64), None)
    progs = []
    for k in range(180):
        e = gen(rng, "BOOLEAN" if k % 2 == 0 else "NUM", 4)
        c = R.RexCompiler(_dev_cols())
        try:
            c.compile(e)
        except R.RexCompileError:
            continue
        progs.append((e, c.prog))
    # targeted datetime programs: jit_civil vs the VM model's civil math
    import types
    from dask_sql_amd.planner.plan import Call, InputRef, SqlType
    for op_name in ("YEAR", "MONTH", "DAY", "EXTRACT_DOY", "EXTRACT_DOW",
                    "EXTRACT_QUARTER", "FLOOR_TO_MONTH", "FLOOR_TO_YEAR",
                    "EXTRACT_CENTURY"):
        e = Call(op_name, [InputRef(0, SqlType("DATE"))], SqlType("DATE"))
        c = R.RexCompiler(_dev_cols())
        c.compile(e)
        progs.append((e, c.prog))
    exprs = []
    sources = []
    kinds = []
    for e, prog in progs:
        src, kind = _emit_source(lib, prog)
        if src is None:
            continue
        i = len(sources)
        sources.append(_devicify_casts(src).replace("j_expr",
                                                    f"j_expr_{i}"))
        kinds.append(kind)
        exprs.append((e, prog))

    td = Path(tempfile.mkdtemp(prefix="dsx_jitdiff_"))
    calls = []
    for i, kind in enumerate(kinds):
        ty = "double" if kind == "d" else "i64"
        calls.append(
            f"    case {i}: {{ {ty} o = 0; int ok = j_expr_{i}(C, r, o); "
            f"printf(\"%d %.17g\\n\", ok, (double)o); break; }}")
    main_src = HARNESS_PRELUDE + "\n".join(sources) + f"""
int main(int argc, char** argv) {{
  int idx = atoi(argv[1]);
  static i64 ic0[{n_rows}], ic1[{n_rows}], dc4[{n_rows}];
  static double fc2[{n_rows}], fc3[{n_rows}];
  static u8 v1[{n_rows}], v3[{n_rows}];
  FILE* f = fopen(argv[2], "rb");
  fread(ic0, 8, {n_rows}, f); fread(ic1, 8, {n_rows}, f);
  fread(fc2, 8, {n_rows}, f); fread(fc3, 8, {n_rows}, f);
  fread(dc4, 8, {n_rows}, f);
  fread(v1, 1, {n_rows}, f); fread(v3, 1, {n_rows}, f);
  fclose(f);
  struct ColsArg C;
  C.ncols = 5;
  C.data[0] = ic0; C.data[1] = ic1; C.data[2] = fc2; C.data[3] = fc3;
  C.data[4] = dc4;
  C.validity[0] = 0; C.validity[1] = v1; C.validity[2] = 0;
  C.validity[3] = v3; C.validity[4] = 0;
  for (i64 r = 0; r < {n_rows}; r++) {{
    switch (idx) {{
{chr(10).join(calls)}
    }}
  }}
  return 0;
}}
"""
    cpath = td / "harness.c"
    cpath.write_text(main_src)
    exe = td / "harness"
    r = subprocess.run(["gcc", "-O1", "-x", "c++", str(cpath), "-o",
                        str(exe), "-lm", "-lstdc++"],
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[:4000]

    data = td / "cols.bin"
    with open(data, "wb") as f:
        f.write(cols[0][0].astype("<i8").tobytes())
        f.write(cols[1][0].astype("<i8").tobytes())
        f.write(cols[2][0].astype("<f8").tobytes())
        f.write(cols[3][0].astype("<f8").tobytes())
        f.write(cols[4][0].astype("<i8").tobytes())
        f.write(cols[1][1].astype("u1").tobytes())
        f.write(cols[3][1].astype("u1").tobytes())

    def run(idx):
        out = subprocess.run([str(exe), str(idx), str(data)],
                             capture_output=True, text=True, timeout=60)
        assert out.returncode == 0
        res = []
        for line in out.stdout.strip().splitlines():
            ok, val = line.split()
            res.append((float(val), ok == "1"))
        return res

    return exprs, kinds, cols, run


def test_jit_codegen_matches_vm(harness):
    exprs, kinds, cols, run = harness
    assert len(exprs) > 60
    checked = 0
    for i, (e, prog) in enumerate(exprs):
        jit_rows = run(i)
        for row, (jval, jok) in enumerate(jit_rows):
            vval, vok = interp(prog, cols, row)
            assert jok == vok, (i, row, e)
            if not vok:
                continue  # value is garbage on both sides when NULL
            checked += 1
            v = float(vval)
            if math.isnan(v):
                assert math.isnan(jval), (i, row, e)
            elif math.isinf(v):
                assert jval == v, (i, row, e)
            else:
                assert abs(jval - v) <= 1e-9 * max(1.0, abs(v)), \
                    (i, row, e, jval, v)
    assert checked > 1500
