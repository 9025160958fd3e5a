"""ctypes binding to libdsxhip.so — the C-ABI boundary (include/dsxhip.h).

This is the ONLY module that touches the HIP library. It fails LOUDLY when
the extension or a GPU is missing: there is no CPU fallback on the product
path.
"""
from __future__ import annotations

import ctypes as ct
from pathlib import Path

import numpy as np

# ---- dtype tags (include/dsxhip.h DsxType) --------------------------------
I64, F64, I32, F32, I8, BOOL8 = 0, 1, 2, 3, 4, 5

_NP_TO_DSX = {
    np.dtype("int64"): I64,
    np.dtype("float64"): F64,
    np.dtype("int32"): I32,
    np.dtype("float32"): F32,
    np.dtype("int8"): I8,
    np.dtype("uint8"): BOOL8,
    np.dtype("bool"): BOOL8,
}
_DSX_TO_NP = {
    I64: np.dtype("int64"),
    F64: np.dtype("float64"),
    I32: np.dtype("int32"),
    F32: np.dtype("float32"),
    I8: np.dtype("int8"),
    BOOL8: np.dtype("uint8"),
}
_DSX_SIZE = {I64: 8, F64: 8, I32: 4, F32: 4, I8: 1, BOOL8: 1}

MAX_PROG = 120
MAX_COLS = 16
MAX_AGGS = 16

# agg ops (include/dsxhip.h DsxAggOp)
AGG_SUM_F64, AGG_SUM_I64, AGG_COUNT = 0, 1, 2
AGG_MIN_F64, AGG_MAX_F64, AGG_MIN_I64, AGG_MAX_I64 = 3, 4, 5, 6

JOIN_INNER, JOIN_LEFT, JOIN_LEFTSEMI, JOIN_LEFTANTI = 0, 1, 2, 3
NULL_IDX = 0xFFFFFFFF


class DsxError(RuntimeError):
    pass


class DsxUnavailable(DsxError):
    """Raised when libdsxhip.so / a GPU is not available. The product path
    must propagate this — never fall back to CPU."""


class _Instr(ct.Structure):
    _fields_ = [("op", ct.c_int32), ("arg0", ct.c_int32), ("imm", ct.c_int64)]


class _Column(ct.Structure):
    _fields_ = [
        ("data", ct.c_void_p),
        ("validity", ct.c_void_p),
        ("len", ct.c_int64),
        ("dtype", ct.c_int32),
    ]


class _KeySpec(ct.Structure):
    _fields_ = [
        ("col", ct.c_int32),
        ("min", ct.c_int64),
        ("range", ct.c_int64),
        ("nullable", ct.c_int32),
        ("mode", ct.c_int32),
    ]


class _AggSpec(ct.Structure):
    _fields_ = [
        ("op", ct.c_int32),
        ("prog_len", ct.c_int32),
        ("prog", _Instr * MAX_PROG),
    ]


_LIB = None


def _load_lib():
    global _LIB
    if _LIB is not None:
        return _LIB
    so = Path(__file__).resolve().parent / "libdsxhip.so"
    if not so.exists():
        raise DsxUnavailable(
            f"HIP extension not built: {so} missing. Run `make -C "
            f"{so.parent / 'csrc'}` (hipcc --offload-arch=gfx950). "
            "The MI355X execution layer has NO CPU fallback."
        )
    lib = ct.CDLL(str(so))
    lib.dsx_last_error.restype = ct.c_char_p
    lib.dsx_ctx_create.argtypes = [ct.c_int, ct.POINTER(ct.c_void_p)]
    _LIB = lib
    return lib


def _check(lib, rc, what):
    if rc != 0:
        raise DsxError(f"{what} failed ({rc}): {lib.dsx_last_error().decode()}")


class DeviceColumn:
    """A device-resident column: data ptr + optional validity + dtype."""

    def __init__(self, rt, data, validity, length, dtype, owner=True,
                 keep_alive=None):
        self.rt = rt
        self.data = data
        self.validity = validity  # device ptr or None
        self.len = length
        self.dtype = dtype
        self._owner = owner
        self._keep_alive = keep_alive  # e.g. torch tensor backing the ptrs

    def c_struct(self):
        # cached: data/validity/len are immutable after construction and
        # ~40 ctypes struct builds per query step measured on the Q3 host
        # overhead profile
        c = getattr(self, "_c_struct", None)
        if c is None:
            c = _Column(self.data, self.validity or None, self.len,
                        self.dtype)
            self._c_struct = c
        return c

    def __del__(self):
        if getattr(self, "_owner", False) and self.rt and self.rt.lib:
            try:
                self.rt.lib.dsx_free(self.rt.ctx, ct.c_void_p(self.data))
                if self.validity:
                    self.rt.lib.dsx_free(self.rt.ctx, ct.c_void_p(self.validity))
            except Exception:
                pass

    def to_numpy(self):
        np_dtype = _DSX_TO_NP[self.dtype]
        out = np.empty(self.len, dtype=np_dtype)
        self.rt._download(self.data, out)
        if self.validity:
            v = np.empty(self.len, dtype=np.uint8)
            self.rt._download(self.validity, v)
            return out, v.astype(bool)
        return out, None


class Runtime:
    """One GPU, one HIP stream, one ctypes session (include/dsxhip.h)."""

    def __init__(self, device_id: int = 0):
        self.lib = _load_lib()
        ctx = ct.c_void_p()
        rc = self.lib.dsx_ctx_create(ct.c_int(device_id), ct.byref(ctx))
        if rc != 0:
            raise DsxUnavailable(
                f"dsx_ctx_create({device_id}) failed: "
                f"{self.lib.dsx_last_error().decode()} — is a GPU visible?"
            )
        self.ctx = ctx
        self.device_id = device_id

    def close(self):
        if getattr(self, "ctx", None):
            self.lib.dsx_ctx_destroy(self.ctx)
            self.ctx = None

    # ---- memory ----------------------------------------------------------
    def _malloc(self, nbytes) -> int:
        p = ct.c_void_p()
        _check(self.lib, self.lib.dsx_malloc(self.ctx, ct.c_int64(max(nbytes, 1)),
                                             ct.byref(p)), "dsx_malloc")
        return p.value

    def _free(self, ptr):
        if ptr:
            self.lib.dsx_free(self.ctx, ct.c_void_p(ptr))

    def _upload_raw(self, arr: np.ndarray) -> int:
        arr = np.ascontiguousarray(arr)
        p = ct.c_void_p()
        _check(
            self.lib,
            self.lib.dsx_upload(self.ctx, arr.ctypes.data_as(ct.c_void_p),
                                ct.c_int64(arr.nbytes), ct.byref(p)),
            "dsx_upload",
        )
        return p.value

    def copy_raw(self, dst_ptr: int, src_ptr: int, nbytes: int):
        """Device→device copy (dsx_copy / hipMemcpyAsync on the lib
        stream); staging into externally-owned buffers (e.g. torch tensors
        for RCCL collectives)."""
        if nbytes:
            _check(self.lib,
                   self.lib.dsx_copy(self.ctx, ct.c_void_p(dst_ptr),
                                     ct.c_void_p(src_ptr),
                                     ct.c_int64(nbytes)), "dsx_copy")

    def _download(self, dev_ptr, out: np.ndarray):
        _check(
            self.lib,
            self.lib.dsx_download(self.ctx, ct.c_void_p(dev_ptr),
                                  out.ctypes.data_as(ct.c_void_p),
                                  ct.c_int64(out.nbytes)),
            "dsx_download",
        )

    _PINNED_MIN = 8 << 20  # below this the pageable path is fine

    def upload_column(self, arr: np.ndarray, validity: np.ndarray | None = None,
                      dtype: int | None = None) -> DeviceColumn:
        if dtype is None:
            dtype = _NP_TO_DSX[arr.dtype]
        data = self._upload_raw_auto(arr)
        vptr = None
        if validity is not None:
            vptr = self._upload_raw_auto(validity.astype(np.uint8))
        return DeviceColumn(self, data, vptr, len(arr), dtype)

    def _upload_raw_auto(self, arr: np.ndarray) -> int:
        """Large buffers go through the pinned staging arena
        (dsx_upload_pinned — chunked memcpy + hipMemcpyAsync overlap);
        small ones take the plain path."""
        arr = np.ascontiguousarray(arr)
        if arr.nbytes < self._PINNED_MIN:
            return self._upload_raw(arr)
        p = ct.c_void_p()
        _check(
            self.lib,
            self.lib.dsx_upload_pinned(self.ctx,
                                       arr.ctypes.data_as(ct.c_void_p),
                                       ct.c_int64(arr.nbytes), ct.byref(p)),
            "dsx_upload_pinned",
        )
        return p.value

    def empty_column(self, n: int, dtype: int, with_validity=False) -> DeviceColumn:
        data = self._malloc(n * _DSX_SIZE[dtype])
        vptr = self._malloc(n) if with_validity else None
        return DeviceColumn(self, data, vptr, n, dtype)

    def synchronize(self):
        _check(self.lib, self.lib.dsx_synchronize(self.ctx), "dsx_synchronize")

    # ---- profiling -------------------------------------------------------
    def prof_enable(self, on=True):
        self.lib.dsx_prof_enable(self.ctx, 1 if on else 0)

    def prof_reset(self):
        self.lib.dsx_prof_reset(self.ctx)

    def prof_get(self) -> dict:
        cap = 32
        names = ((ct.c_char * 32) * cap)()
        ms = (ct.c_double * cap)()
        launches = (ct.c_int64 * cap)()
        n = self.lib.dsx_prof_get(self.ctx, names, ms, launches, cap)
        return {
            names[i].value.decode(): {"ms": ms[i], "launches": launches[i]}
            for i in range(n)
        }

    # ---- helpers ---------------------------------------------------------
    # stack effect per opcode family (VM slots are 6 named registers —
    # dsxhip.hip VmStack; exceeding them would silently alias slot 5)
    _PUSH_OPS = {1, 2, 3, 4}          # COL, LIT_*
    _BIN_OPS = set(range(10, 42))     # arith/cmp/AND/OR
    _SELECT_OP = 60
    _MAX_DEPTH = 8

    @classmethod
    def _check_depth(cls, instrs):
        depth = 0
        peak = 0
        for op, _, _ in instrs:
            if op in cls._PUSH_OPS:
                depth += 1
            elif op in cls._BIN_OPS and op != 42:  # NOT(42) is unary
                depth -= 1
            elif op == cls._SELECT_OP:
                depth -= 2
            peak = max(peak, depth)
        if peak > cls._MAX_DEPTH:
            raise DsxError(
                f"expression too deep for the VM register stack "
                f"({peak} > {cls._MAX_DEPTH}); split the expression "
                "(e.g. nested CASE) into projection steps")

    @staticmethod
    def make_prog(instrs) -> tuple:
        """instrs: list of (op, arg0, imm) where imm may be float/int.
        Cached by content — the ctypes arrays are immutable after build and
        the C side copies them, so reuse across steps is safe."""
        return Runtime._make_prog_cached(tuple(instrs))

    @staticmethod
    def _make_prog_cached(instrs) -> tuple:
        cache = Runtime._prog_cache
        hit = cache.get(instrs)
        if hit is not None:
            return hit
        if len(instrs) > MAX_PROG:
            raise DsxError(f"program too long ({len(instrs)})")
        Runtime._check_depth(instrs)
        arr = (_Instr * max(len(instrs), 1))()
        for i, (op, arg0, imm) in enumerate(instrs):
            arr[i].op = op
            arr[i].arg0 = arg0
            if isinstance(imm, float):
                arr[i].imm = int.from_bytes(
                    np.float64(imm).tobytes(), "little", signed=True
                )
            else:
                arr[i].imm = int(imm)
        if len(cache) > 4096:
            cache.clear()
        cache[instrs] = (arr, len(instrs))
        return arr, len(instrs)

    _prog_cache: dict = {}

    @staticmethod
    def _cols_array(cols):
        arr = (_Column * max(len(cols), 1))()
        for i, c in enumerate(cols):
            arr[i] = c.c_struct()
        return arr

    # ---- ops -------------------------------------------------------------
    def eval(self, prog, cols, n, out_dtype, with_validity=True) -> DeviceColumn:
        parr, plen = prog
        out = self.empty_column(n, out_dtype, with_validity)
        _check(
            self.lib,
            self.lib.dsx_eval(self.ctx, parr, ct.c_int(plen),
                              self._cols_array(cols), ct.c_int(len(cols)),
                              ct.c_int64(n), ct.c_void_p(out.data),
                              ct.c_void_p(out.validity) if out.validity else None,
                              ct.c_int32(out_dtype)),
            "dsx_eval",
        )
        return out

    def filter(self, prog, cols, n):
        """Returns (sel_device_ptr, count) — ordered selection vector."""
        parr, plen = prog
        sel = ct.c_void_p()
        count = ct.c_int64()
        _check(
            self.lib,
            self.lib.dsx_filter(self.ctx, parr, ct.c_int(plen),
                                self._cols_array(cols), ct.c_int(len(cols)),
                                ct.c_int64(n), ct.byref(sel), ct.byref(count)),
            "dsx_filter",
        )
        return sel.value, count.value

    def gather_into(self, col: DeviceColumn, sel_ptr, n_sel, out_ptr: int):
        """Gather into a caller-owned device buffer (e.g. a torch tensor's
        storage, for the RCCL shuffle staging — SURVEY §8e)."""
        _check(
            self.lib,
            self.lib.dsx_gather(self.ctx, ct.byref(col.c_struct()),
                                ct.c_void_p(sel_ptr), ct.c_int64(n_sel),
                                ct.c_void_p(out_ptr), None),
            "dsx_gather",
        )

    def wrap_devptr(self, ptr: int, n: int, dtype: int,
                    keep_alive=None) -> DeviceColumn:
        """Wrap an externally-owned device buffer (e.g. torch tensor)."""
        return DeviceColumn(self, ptr, None, n, dtype, owner=False,
                            keep_alive=keep_alive)

    def concat_columns(self, cols, dtype) -> DeviceColumn:
        """Concatenate device columns of one dtype (UNION ALL — reference
        dd.concat). Validity kept if any input has one (absent = all-1)."""
        total = sum(c.len for c in cols)
        need_valid = any(c.validity for c in cols)
        out = self.empty_column(total, dtype, need_valid)
        sz = _DSX_SIZE[dtype]
        off = 0
        for c in cols:
            assert c.dtype == dtype
            if c.len:
                _check(self.lib,
                       self.lib.dsx_copy(self.ctx,
                                         ct.c_void_p(out.data + off * sz),
                                         ct.c_void_p(c.data),
                                         ct.c_int64(c.len * sz)), "dsx_copy")
                if need_valid:
                    if c.validity:
                        _check(self.lib, self.lib.dsx_copy(
                            self.ctx, ct.c_void_p(out.validity + off),
                            ct.c_void_p(c.validity), ct.c_int64(c.len)),
                            "dsx_copy")
                    else:
                        _check(self.lib, self.lib.dsx_memset(
                            self.ctx, ct.c_void_p(out.validity + off),
                            ct.c_int(1), ct.c_int64(c.len)), "dsx_memset")
            off += c.len
        return out

    def scatter_rows(self, col: DeviceColumn, sel_ptr, n_sel, n_out,
                     with_validity=True) -> DeviceColumn:
        """out[sel[i]] = col[i] (inverse of gather; window/join-back
        placement). Rows not covered by sel come out NULL."""
        out = self.empty_column(n_out, col.dtype, with_validity)
        if out.validity:
            _check(self.lib,
                   self.lib.dsx_memset(self.ctx, ct.c_void_p(out.validity),
                                       ct.c_int(0), ct.c_int64(n_out)),
                   "dsx_memset")
        _check(
            self.lib,
            self.lib.dsx_scatter_rows(self.ctx, ct.byref(col.c_struct()),
                                      ct.c_void_p(sel_ptr),
                                      ct.c_int64(n_sel), ct.c_int64(n_out),
                                      ct.c_void_p(out.data),
                                      ct.c_void_p(out.validity)
                                      if out.validity else None),
            "dsx_scatter_rows",
        )
        return out

    def gather(self, col: DeviceColumn, sel_ptr, n_sel,
               force_validity=False) -> DeviceColumn:
        need_valid = force_validity or bool(col.validity)
        out = self.empty_column(n_sel, col.dtype, need_valid)
        _check(
            self.lib,
            self.lib.dsx_gather(self.ctx, ct.byref(col.c_struct()),
                                ct.c_void_p(sel_ptr), ct.c_int64(n_sel),
                                ct.c_void_p(out.data),
                                ct.c_void_p(out.validity) if out.validity else None),
            "dsx_gather",
        )
        return out

    def minmax_i64(self, col: DeviceColumn):
        mn = ct.c_int64()
        mx = ct.c_int64()
        nn = ct.c_int64()
        _check(
            self.lib,
            self.lib.dsx_minmax_i64(self.ctx, ct.byref(col.c_struct()),
                                    ct.byref(mn), ct.byref(mx), ct.byref(nn)),
            "dsx_minmax_i64",
        )
        return mn.value, mx.value, nn.value

    def keypack(self, cols, keyspecs, n) -> tuple:
        """keyspecs: list of (col_idx, min, range, nullable).
        Returns (codes DeviceColumn(u64-as-i64), key_space)."""
        ks = (_KeySpec * len(keyspecs))()
        space = 1
        for i, spec in enumerate(keyspecs):
            ci, mn, rng, nullable = spec[:4]
            ks[i].mode = spec[4] if len(spec) > 4 else 0
            ks[i].col = ci
            ks[i].min = mn
            ks[i].range = rng
            ks[i].nullable = 1 if nullable else 0
            space *= rng + (1 if nullable else 0)
        out = self.empty_column(n, I64, False)
        _check(
            self.lib,
            self.lib.dsx_keypack(self.ctx, self._cols_array(cols),
                                 ct.c_int(len(cols)), ks, ct.c_int(len(keyspecs)),
                                 ct.c_int64(n), ct.c_void_p(out.data)),
            "dsx_keypack",
        )
        return out, space

    def hash_build(self, codes: DeviceColumn, validity_ptr=None, code_max=0):
        t = ct.c_void_p()
        _check(
            self.lib,
            self.lib.dsx_hash_build(self.ctx, ct.c_void_p(codes.data),
                                    ct.c_void_p(validity_ptr) if validity_ptr else None,
                                    ct.c_int64(codes.len),
                                    ct.c_uint64(code_max), ct.byref(t)),
            "dsx_hash_build",
        )
        return t

    def hash_probe(self, table, codes: DeviceColumn, join_type,
                   validity_ptr=None, mark_matched=False):
        p = ct.c_void_p()
        b = ct.c_void_p()
        count = ct.c_int64()
        _check(
            self.lib,
            self.lib.dsx_hash_probe(self.ctx, table, ct.c_void_p(codes.data),
                                    ct.c_void_p(validity_ptr) if validity_ptr else None,
                                    ct.c_int64(codes.len), ct.c_int(join_type),
                                    ct.c_int(1 if mark_matched else 0),
                                    ct.byref(p), ct.byref(b), ct.byref(count)),
            "dsx_hash_probe",
        )
        return p.value, b.value, count.value

    def hash_unmatched(self, table):
        b = ct.c_void_p()
        count = ct.c_int64()
        _check(
            self.lib,
            self.lib.dsx_hash_unmatched(self.ctx, table, ct.byref(b),
                                        ct.byref(count)),
            "dsx_hash_unmatched",
        )
        return b.value, count.value

    def filter_cols(self, prog, cols, n, mats):
        """Fused filter + materialization (dsx_filter_cols): returns
        ([DeviceColumn per mat], n_out)."""
        parr, plen = prog
        nm = len(mats)
        datas = (ct.c_void_p * max(nm, 1))()
        valids = (ct.c_void_p * max(nm, 1))()
        count = ct.c_int64()
        _check(
            self.lib,
            self.lib.dsx_filter_cols(
                self.ctx, parr, ct.c_int(plen), self._cols_array(cols),
                ct.c_int(len(cols)), ct.c_int64(n),
                self._cols_array(mats), ct.c_int(nm), datas, valids,
                ct.byref(count)),
            "dsx_filter_cols",
        )
        nr = count.value
        return [DeviceColumn(self, datas[i], valids[i] or None, nr,
                             m.dtype, owner=True)
                for i, m in enumerate(mats)], nr

    def hash_probe_cols(self, table, codes: DeviceColumn, join_type,
                        validity_ptr, pcols, bcols, force_build_validity):
        """Fused probe-emit + materialization (dsx_hash_probe_cols): returns
        ([DeviceColumn] ordered pcols then bcols, n_out)."""
        ncols = len(pcols) + len(bcols)
        datas = (ct.c_void_p * max(ncols, 1))()
        valids = (ct.c_void_p * max(ncols, 1))()
        count = ct.c_int64()
        _check(
            self.lib,
            self.lib.dsx_hash_probe_cols(
                self.ctx, table, ct.c_void_p(codes.data),
                ct.c_void_p(validity_ptr) if validity_ptr else None,
                ct.c_int64(codes.len), ct.c_int(join_type),
                self._cols_array(pcols), ct.c_int(len(pcols)),
                self._cols_array(bcols), ct.c_int(len(bcols)),
                ct.c_int(1 if force_build_validity else 0),
                datas, valids, ct.byref(count)),
            "dsx_hash_probe_cols",
        )
        n = count.value
        out = []
        for i, src in enumerate(list(pcols) + list(bcols)):
            out.append(DeviceColumn(self, datas[i], valids[i] or None, n,
                                    src.dtype, owner=True))
        return out, n

    def radix_join(self, bcols, n_build, keyspecs_b, keyspecs_p, bpred_prog,
                   pcols, n_probe, join_type, out_specs):
        """Radix-partitioned equijoin (dsx_radix_join): out_specs =
        [(side, col_idx, need_valid)] with side 0=probe/1=build and col_idx
        into that side's cols array. Returns ([DeviceColumn], n) or None on
        bucket overflow / unsupported shape (caller falls back to the
        flat-table join)."""
        nk = len(keyspecs_b)
        kb = (_KeySpec * nk)()
        kp = (_KeySpec * nk)()
        for i, (sb, sp) in enumerate(zip(keyspecs_b, keyspecs_p)):
            for arr, spec in ((kb, sb), (kp, sp)):
                arr[i].col = spec[0]
                arr[i].min = spec[1]
                arr[i].range = spec[2]
                arr[i].nullable = 1 if spec[3] else 0
                arr[i].mode = spec[4] if len(spec) > 4 else 0
        n_out = len(out_specs)
        side_a = (ct.c_int32 * n_out)(*[s[0] for s in out_specs])
        col_a = (ct.c_int32 * n_out)(*[s[1] for s in out_specs])
        nv_a = (ct.c_int32 * n_out)(*[1 if s[2] else 0 for s in out_specs])
        datas = (ct.c_void_p * n_out)()
        valids = (ct.c_void_p * n_out)()
        count = ct.c_int64()
        bp = None
        bl = 0
        if bpred_prog:
            bp, bl = self.make_prog(bpred_prog)
        rc = self.lib.dsx_radix_join(
            self.ctx, self._cols_array(bcols), ct.c_int(len(bcols)),
            ct.c_int64(n_build), kb, kp, ct.c_int(nk),
            bp, ct.c_int(bl),
            self._cols_array(pcols), ct.c_int(len(pcols)),
            ct.c_int64(n_probe), ct.c_int(join_type),
            side_a, col_a, nv_a, ct.c_int(n_out),
            datas, valids, ct.byref(count))
        if rc in (-5, -6):
            return None  # skew overflow / degenerate: flat-table fallback
        _check(self.lib, rc, "dsx_radix_join")
        n = count.value
        out = []
        for i, (side, ci, _) in enumerate(out_specs):
            src = bcols[ci] if side else pcols[ci]
            out.append(DeviceColumn(self, datas[i], valids[i] or None, n,
                                    src.dtype, owner=True))
        return out, n

    def sort_perm(self, cols, keyspecs, n):
        """Device ORDER BY permutation (dsx_sort_perm): keyspecs =
        (col, min, range, nullable, mode) with mode bit1=DESC,
        bit2=NULLS LAST, passed in REVERSED significance order. Returns a
        DeviceColumn of u32 row ids or None (unsupported/skew → host)."""
        nk = len(keyspecs)
        ks = (_KeySpec * nk)()
        for i, spec in enumerate(keyspecs):
            ks[i].col = spec[0]
            ks[i].min = spec[1]
            ks[i].range = spec[2]
            ks[i].nullable = 1 if spec[3] else 0
            ks[i].mode = spec[4]
        perm = ct.c_void_p()
        rc = self.lib.dsx_sort_perm(self.ctx, self._cols_array(cols),
                                    ct.c_int(len(cols)), ks, ct.c_int(nk),
                                    ct.c_int64(n), ct.byref(perm))
        if rc in (-3, -4, -6):
            return None
        _check(self.lib, rc, "dsx_sort_perm")
        return DeviceColumn(self, perm.value, None, n, I32, owner=True)

    WIN_FUNCS = {"row_number": 0, "rank": 1, "dense_rank": 2, "lag": 3,
                 "lead": 4, "first_value": 5, "sum": 6, "count": 7,
                 "min": 8, "max": 9, "avg": 10}

    def window_ordered(self, perm, n, pcode_sorted, fcode_sorted, func,
                       v_col, offset, default_bits, has_default, out_dtype,
                       want_valid):
        """Device ordered window frame (dsx_window_ordered)."""
        d = ct.c_void_p()
        vv = ct.c_void_p()
        vstruct = None
        if v_col is not None:
            vstruct = v_col.c_struct()
        _check(
            self.lib,
            self.lib.dsx_window_ordered(
                self.ctx, ct.c_void_p(perm.data), ct.c_int64(n),
                ct.c_void_p(pcode_sorted.data),
                ct.c_void_p(fcode_sorted.data) if fcode_sorted is not None
                else None,
                ct.c_int(self.WIN_FUNCS[func]),
                ct.byref(vstruct) if vstruct is not None else None,
                ct.c_int64(offset), ct.c_int64(default_bits),
                ct.c_int(1 if has_default else 0), ct.c_int(out_dtype),
                ct.byref(d), ct.byref(vv),
                ct.c_int(1 if want_valid else 0)),
            "dsx_window_ordered",
        )
        return DeviceColumn(self, d.value, vv.value or None, n, out_dtype,
                            owner=True)

    def hash_table_free(self, table):
        self.lib.dsx_hash_table_free(table)

    def hash_groupby(self, cols, n, keyspecs, pred_prog, agg_specs):
        # the C-side hist cache keys on device pointers with pool-free
        # eviction; externally-backed key columns (torch/RCCL staging)
        # recycle pointers outside that hook — disable for them
        unsafe = any(
            getattr(cols[spec[0]], "_keep_alive", None) is not None
            and not getattr(cols[spec[0]], "_owner", True)
            for spec in keyspecs)
        if unsafe:
            self.lib.dsx_gb_hist_cache_enable(self.ctx, 0)
        try:
            return self._hash_groupby(cols, n, keyspecs, pred_prog,
                                      agg_specs)
        finally:
            if unsafe:
                self.lib.dsx_gb_hist_cache_enable(self.ctx, 1)

    def _hash_groupby(self, cols, n, keyspecs, pred_prog, agg_specs):
        """keyspecs: list of (col_idx, min, range, nullable) — key pack fused
        in-kernel. agg_specs: list of (agg_op, prog). Returns device pointers
        (out_codes, out_vals [naggs][G], out_counts [naggs][G], n_groups)."""
        ks = (_KeySpec * max(len(keyspecs), 1))()
        for i, spec in enumerate(keyspecs):
            ci, mn, rng, nullable = spec[:4]
            ks[i].mode = spec[4] if len(spec) > 4 else 0
            ks[i].col = ci
            ks[i].min = mn
            ks[i].range = rng
            ks[i].nullable = 1 if nullable else 0
        aggs = (_AggSpec * max(len(agg_specs), 1))()
        for i, (op, prog) in enumerate(agg_specs):
            parr, plen = prog
            aggs[i].op = op
            aggs[i].prog_len = plen
            ct.memmove(aggs[i].prog, parr, plen * ct.sizeof(_Instr))
        if pred_prog is not None:
            pparr, pplen = pred_prog
        else:
            pparr, pplen = (_Instr * 1)(), 0
        oc = ct.c_void_p()
        ov = ct.c_void_p()
        on = ct.c_void_p()
        og = ct.c_int64()
        _check(
            self.lib,
            self.lib.dsx_hash_groupby(
                self.ctx, self._cols_array(cols), ct.c_int(len(cols)),
                ct.c_int64(n), ks, ct.c_int(len(keyspecs)), pparr,
                ct.c_int(pplen), aggs, ct.c_int(len(agg_specs)),
                ct.byref(oc), ct.byref(ov), ct.byref(on), ct.byref(og)),
            "dsx_hash_groupby",
        )
        return oc.value, ov.value, on.value, og.value

    def partition(self, codes: DeviceColumn, nbuckets, validity_ptr=None):
        """Returns (sel DeviceColumn-like ptr wrapped, offsets np.ndarray)."""
        sel = self.empty_column(codes.len, I32, False)  # u32 indices
        offsets = np.zeros(nbuckets + 1, dtype=np.int64)
        _check(
            self.lib,
            self.lib.dsx_partition(self.ctx, ct.c_void_p(codes.data),
                                   ct.c_void_p(validity_ptr) if validity_ptr else None,
                                   ct.c_int64(codes.len), ct.c_int(nbuckets),
                                   ct.c_void_p(sel.data),
                                   offsets.ctypes.data_as(ct.POINTER(ct.c_int64))),
            "dsx_partition",
        )
        return sel, offsets

    def wrap_sel(self, sel_ptr, count) -> DeviceColumn:
        """Wrap a library-allocated u32 selection vector."""
        return DeviceColumn(self, sel_ptr, None, count, I32, owner=True)
