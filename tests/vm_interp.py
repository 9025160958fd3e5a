"""CPU interpreter of the rex VM, slot-for-slot faithful to the device
interpreter (csrc/dsxhip.hip vm_eval switch) — test infrastructure only.
Each stack slot is (value, valid); semantics (Kleene AND/OR, guarded
integer division, trunc float→int, validity propagation) mirror the .hip
cases so CPU differential tests pin the compiler without a GPU."""
import math
import struct

from dask_sql_amd.physical import rex as R


def _tdiv(a, b):
    q = abs(a) // abs(b)
    return q if (a >= 0) == (b >= 0) else -q


def _civil(days):
    import numpy as np
    d = np.datetime64(int(days), "D").astype(object)
    return d.year, d.month, d.day


def _i64(v):
    # wrap to signed 64-bit like the device's int64_t arithmetic
    v &= (1 << 64) - 1
    return v - (1 << 64) if v >= (1 << 63) else v


def interp(prog, cols, row):
    """cols: list of (values, validity|None) arrays; row: index.
    Returns (value, valid) of the expression for that row."""
    st = []

    def push(v, ok):
        st.append((v, ok))

    def pop2():
        b, bv = st.pop()
        a, av = st.pop()
        return a, av, b, bv

    def pop1():
        return st.pop()

    for op, arg0, imm in prog:
        if op == R.OP_COL:
            vals, valid = cols[arg0]
            ok = True if valid is None else bool(valid[row])
            v = vals[row]
            push(float(v) if hasattr(v, "dtype") and "float" in str(v.dtype)
                 else (float(v) if isinstance(v, float) else int(v)), ok)
        elif op == R.OP_LIT_F64:
            push(float(imm), True)
        elif op == R.OP_LIT_I64:
            push(int(imm), True)
        elif op == R.OP_LIT_NULL:
            push(0, False)
        elif op == R.OP_ADD_I64:
            a, av, b, bv = pop2()
            push(_i64(int(a) + int(b)), av and bv)
        elif op == R.OP_SUB_I64:
            a, av, b, bv = pop2()
            push(_i64(int(a) - int(b)), av and bv)
        elif op == R.OP_MUL_I64:
            a, av, b, bv = pop2()
            push(_i64(int(a) * int(b)), av and bv)
        elif op == R.OP_DIV_I64:
            a, av, b, bv = pop2()
            push(_tdiv(int(a), int(b)) if b else 0, av and bv)
        elif op == R.OP_MOD_I64:
            a, av, b, bv = pop2()
            push(int(a) - _tdiv(int(a), int(b)) * int(b) if b else 0,
                 av and bv)
        elif op == R.OP_FLOORMOD_I64:
            a, av, b, bv = pop2()
            push(((int(a) % int(b)) + int(b)) % int(b) if b else 0,
                 av and bv)
        elif op == R.OP_ADD_F64:
            a, av, b, bv = pop2()
            push(float(a) + float(b), av and bv)
        elif op == R.OP_SUB_F64:
            a, av, b, bv = pop2()
            push(float(a) - float(b), av and bv)
        elif op == R.OP_MUL_F64:
            a, av, b, bv = pop2()
            push(float(a) * float(b), av and bv)
        elif op == R.OP_DIV_F64:
            a, av, b, bv = pop2()
            if float(b) == 0.0:
                q = math.nan if float(a) == 0.0 else \
                    math.copysign(math.inf, float(a)) * \
                    math.copysign(1.0, float(b))
            else:
                q = float(a) / float(b)
            push(q, av and bv)
        elif op in (R.OP_LT_I64, R.OP_LT_F64):
            a, av, b, bv = pop2()
            push(1 if a < b else 0, av and bv)
        elif op in (R.OP_LE_I64, R.OP_LE_F64):
            a, av, b, bv = pop2()
            push(1 if a <= b else 0, av and bv)
        elif op in (R.OP_GT_I64, R.OP_GT_F64):
            a, av, b, bv = pop2()
            push(1 if a > b else 0, av and bv)
        elif op in (R.OP_GE_I64, R.OP_GE_F64):
            a, av, b, bv = pop2()
            push(1 if a >= b else 0, av and bv)
        elif op in (R.OP_EQ_I64, R.OP_EQ_F64):
            a, av, b, bv = pop2()
            push(1 if a == b else 0, av and bv)
        elif op in (R.OP_NE_I64, R.OP_NE_F64):
            a, av, b, bv = pop2()
            push(1 if a != b else 0, av and bv)
        elif op == R.OP_AND:
            a, av, b, bv = pop2()
            fa, fb = av and a == 0, bv and b == 0
            false_wins = fa or fb
            push(1 if (not false_wins and av and bv) else 0,
                 false_wins or (av and bv))
        elif op == R.OP_OR:
            a, av, b, bv = pop2()
            ta, tb = av and a != 0, bv and b != 0
            true_wins = ta or tb
            push(1 if true_wins else 0, true_wins or (av and bv))
        elif op == R.OP_NOT:
            a, av = pop1()
            push(0 if a else 1, av)
        elif op == R.OP_IS_NULL:
            a, av = pop1()
            push(0 if av else 1, True)
        elif op == R.OP_IS_NOT_NULL:
            a, av = pop1()
            push(1 if av else 0, True)
        elif op == R.OP_I64_TO_F64:
            a, av = pop1()
            push(float(int(a)), av)
        elif op == R.OP_F64_TO_I64:
            a, av = pop1()
            f = float(a)
            if math.isnan(f):
                v = 0
            elif f >= 2.0 ** 63:
                v = (1 << 63) - 1   # hardware cvt saturates
            elif f <= -(2.0 ** 63):
                v = -(1 << 63)
            else:
                v = _i64(int(f))
            push(v, av)
        elif op == R.OP_BITS_F64:
            a, av = pop1()
            push(struct.unpack("<d", struct.pack("<q", int(a)))[0], av)
        elif op == R.OP_SELECT:
            b, bv = st.pop()
            a, av = st.pop()
            c, cv = st.pop()
            take = cv and c != 0
            push(a if take else b, av if take else bv)
        elif op == R.OP_NEG_F64:
            a, av = pop1()
            push(-float(a), av)
        elif op == R.OP_NEG_I64:
            a, av = pop1()
            push(_i64(-int(a)), av)
        elif op == R.OP_SQRT_F64:
            a, av = pop1()
            f = float(a)
            push(math.nan if f < 0 else math.sqrt(f), av)
        elif op == R.OP_ABS_I64:
            a, av = pop1()
            push(abs(int(a)), av)
        elif op == R.OP_ABS_F64:
            a, av = pop1()
            push(abs(float(a)), av)
        elif op == R.OP_FLOOR_F64:
            a, av = pop1()
            push(math.floor(float(a)) if math.isfinite(float(a))
                 else float(a), av)
        elif op == R.OP_CEIL_F64:
            a, av = pop1()
            push(math.ceil(float(a)) if math.isfinite(float(a))
                 else float(a), av)
        elif op == R.OP_RINT_F64:
            a, av = pop1()
            f = float(a)
            if math.isfinite(f):
                fl = math.floor(f)
                d = f - fl
                if d > 0.5:
                    f = fl + 1
                elif d < 0.5:
                    f = fl
                else:
                    f = fl if fl % 2 == 0 else fl + 1
            push(float(f), av)
        elif op == R.OP_EXP_F64:
            a, av = pop1()
            try:
                push(math.exp(float(a)), av)
            except OverflowError:
                push(math.inf, av)
        elif op == R.OP_LN_F64:
            a, av = pop1()
            f = float(a)
            push(math.nan if f < 0 else
                 (-math.inf if f == 0 else math.log(f)), av)
        elif op == R.OP_POW_F64:
            a, av, b, bv = pop2()
            try:
                r = math.pow(float(a), float(b))
            except (OverflowError, ValueError):
                r = math.nan
            push(r, av and bv)
        elif op == R.OP_SIN_F64:
            a, av = pop1()
            push(math.sin(float(a)), av)
        elif op == R.OP_COS_F64:
            a, av = pop1()
            push(math.cos(float(a)), av)
        elif op == R.OP_TAN_F64:
            a, av = pop1()
            push(math.tan(float(a)), av)
        elif op == R.OP_ASIN_F64:
            a, av = pop1()
            f = float(a)
            push(math.asin(f) if -1.0 <= f <= 1.0 else math.nan, av)
        elif op == R.OP_ACOS_F64:
            a, av = pop1()
            f = float(a)
            push(math.acos(f) if -1.0 <= f <= 1.0 else math.nan, av)
        elif op == R.OP_ATAN_F64:
            a, av = pop1()
            push(math.atan(float(a)), av)
        elif op == R.OP_ATAN2_F64:
            a, av, b, bv = pop2()
            push(math.atan2(float(a), float(b)), av and bv)
        elif op == R.OP_YEAR:
            a, av = pop1()
            push(_civil(a)[0], av)
        elif op == R.OP_MONTH:
            a, av = pop1()
            push(_civil(a)[1], av)
        elif op == R.OP_DAY:
            a, av = pop1()
            push(_civil(a)[2], av)
        else:
            raise AssertionError(f"opcode {op} not modeled")
    assert len(st) == 1, f"stack depth {len(st)} at end"
    return st[0]
