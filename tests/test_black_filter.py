"""Black (generic-expression) pushdown filters — SURVEY §8(f) row 4.

The reference evaluates non-white predicates by decoding the referenced
columns and running the expression per row
(ObPhysicalFilterExecutor::filter_batch, ob_pushdown_filter.cpp:2066),
with the dict optimization of evaluating the expression once per dict
entry into a ref bitmap (ob_dict_decoder.cpp:1481-1561). The engine's
restatement is a bounded postfix bytecode on the filter leaf (obx.h
OBX_BX_*). Three implementations: the oracle (C), the device kernels
(HIP), and the NAIVE PYTHON evaluator below — the tests pin all three to
each other.
"""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oceanbase_amd import abi, oracle  # noqa: E402

M64 = (1 << 64) - 1


def py_black_eval(prog, consts, vals, nulls):
    """Independent three-valued evaluator (wrap-mod-2^64 arithmetic)."""
    st = []

    def s64(x):
        x &= M64
        return x - (1 << 64) if x >= (1 << 63) else x

    for op in prog:
        if op < 0x40:
            st.append((vals[op], nulls[op]))
        elif op < 0x50:
            st.append((consts[op & 0xF], False))
        elif op == 0x54:
            v, n = st.pop()
            st.append((s64(-v), n))
        elif op == 0x72:
            v, n = st.pop()
            st.append((v if n else (0 if v else 1), n))
        else:
            b, nb = st.pop()
            a, na = st.pop()
            rn = na or nb
            r = 0
            if op == 0x50:
                r = s64(a + b)
            elif op == 0x51:
                r = s64(a - b)
            elif op == 0x52:
                r = s64(a * b)
            elif op == 0x53:
                if b == 0:
                    rn = True
                elif a == -(1 << 63) and b == -1:
                    r = a
                else:
                    r = int(a / b) if (a < 0) != (b < 0) and a % b else a // b
            elif op == 0x55:  # MOD, C truncated-division remainder
                if b == 0:
                    rn = True
                elif a == -(1 << 63) and b == -1:
                    r = 0
                else:
                    q = int(a / b) if (a < 0) != (b < 0) and a % b else a // b
                    r = s64(a - q * b)
            elif 0x60 <= op <= 0x65:
                r = int({0x60: a < b, 0x61: a <= b, 0x62: a > b,
                         0x63: a >= b, 0x64: a == b, 0x65: a != b}[op])
            elif op == 0x70:
                if (not na and not a) or (not nb and not b):
                    r, rn = 0, False
                elif rn:
                    r = 0
                else:
                    r = 1
            elif op == 0x71:
                if (not na and a) or (not nb and b):
                    r, rn = 1, False
                elif rn:
                    r = 0
                else:
                    r = 0
            st.append((r, rn))
    v, n = st[0]
    return (not n) and v != 0


def _mk_block(cols_vals, encs, nulls=None):
    specs = [(abi.T_INT, 0, 19, 8)] * len(cols_vals)
    schema = oracle.make_schema(specs)
    arrays = [np.asarray(v, dtype=np.int64).view(np.uint8)
              for v in cols_vals]
    nb = None
    if nulls:
        nb = []
        rows = len(cols_vals[0])
        for c in range(len(cols_vals)):
            if nulls.get(c):
                bm = np.zeros((rows + 7) // 8, dtype=np.uint8)
                for r in nulls[c]:
                    bm[r // 8] |= 1 << (r % 8)
                nb.append(bm)
            else:
                nb.append(None)
    blk = oracle.encode_block(schema, arrays, encs, nb)
    return schema, blk


def _expected(cols_vals, nulls, prog, consts, bcols):
    rows = len(cols_vals[0])
    out = []
    for r in range(rows):
        vals = [int(cols_vals[c][r]) for c in bcols]
        nus = [bool(nulls and r in nulls.get(c, ())) for c in bcols]
        out.append(py_black_eval(prog, consts, vals, nus))
    return out


def _filter_rows(schema, blk, filt, n_cols):
    import numpy as _np
    rows = int(_np.frombuffer(bytes(blk[16:20]), dtype=_np.uint32)[0])
    bits, pc = oracle.filter_block(schema, n_cols, blk, filt)
    got = [(bits[r >> 3] >> (r & 7)) & 1 == 1 for r in range(rows)]
    return got, pc


CASES = [
    # price * disc > 2000000  (two-column product)
    dict(bcols=[0, 1],
         prog=[abi.BX_COL | 0, abi.BX_COL | 1, abi.BX_MUL,
               abi.BX_CONST | 0, abi.BX_GT],
         consts=[2_000_000]),
    # (a + c0) / c1 != a  with division
    dict(bcols=[0],
         prog=[abi.BX_COL | 0, abi.BX_CONST | 0, abi.BX_ADD,
               abi.BX_CONST | 1, abi.BX_DIV, abi.BX_COL | 0, abi.BX_NE],
         consts=[37, 5]),
    # NOT (a < c0 AND b > c1)  (boolean structure)
    dict(bcols=[0, 1],
         prog=[abi.BX_COL | 0, abi.BX_CONST | 0, abi.BX_LT,
               abi.BX_COL | 1, abi.BX_CONST | 1, abi.BX_GT,
               abi.BX_AND, abi.BX_NOT],
         consts=[500, 3]),
    # -a * b = division-free mixed signs
    dict(bcols=[0, 1],
         prog=[abi.BX_COL | 0, abi.BX_NEG, abi.BX_COL | 1, abi.BX_MUL,
               abi.BX_CONST | 0, abi.BX_LE],
         consts=[-12345]),
    # x / 0 -> NULL -> row drops (div by zero via const)
    dict(bcols=[0],
         prog=[abi.BX_COL | 0, abi.BX_CONST | 0, abi.BX_DIV,
               abi.BX_CONST | 1, abi.BX_GT],
         consts=[0, 1]),
]


@pytest.mark.parametrize("case", range(len(CASES)))
@pytest.mark.parametrize("encs", [
    [abi.ENC_RAW, abi.ENC_RAW],
    [abi.ENC_DICT, abi.ENC_INT_DIFF],
    [abi.ENC_RLE, abi.ENC_DICT],
])
def test_black_filter_block_cpu(case, encs):
    c = CASES[case]
    rng = np.random.default_rng(case * 10 + len(encs))
    rows = 700
    a = (rng.choice([3, 11, 999, 1500], rows) if encs[0] != abi.ENC_RLE
         else np.repeat(rng.choice([5, 70, 900], 20), 35)[:rows])
    b = rng.integers(0, 8, rows) + 1000
    nulls = {0: set(rng.choice(rows, 30, replace=False).tolist())}
    schema, blk = _mk_block([a, b], encs, nulls)
    filt = abi.make_filter([dict(op=abi.OP_BLACK, bcols=c["bcols"],
                                 bprog=c["prog"], bconst=c["consts"])])
    got, pc = _filter_rows(schema, blk, filt, 2)
    exp = _expected([a, b], nulls, c["prog"], c["consts"], c["bcols"])
    assert got == exp, f"case {case} encs {encs}"
    assert pc == sum(exp)


def test_black_with_white_or_combine_cpu():
    rng = np.random.default_rng(4)
    rows = 900
    a = rng.integers(0, 2000, rows)
    b = rng.integers(0, 50, rows)
    schema, blk = _mk_block([a, b], [abi.ENC_RAW, abi.ENC_DICT])
    prog = [abi.BX_COL | 0, abi.BX_COL | 1, abi.BX_MUL,
            abi.BX_CONST | 0, abi.BX_GE]
    filt = abi.make_filter(
        [dict(op=abi.OP_BLACK, bcols=[0, 1], bprog=prog, bconst=[40000]),
         dict(col=1, op=abi.OP_LT, lo=5)],
        prog=[0, 1, abi.TOK_OR])
    got, pc = _filter_rows(schema, blk, filt, 2)
    black = _expected([a, b], None, prog, [40000], [0, 1])
    exp = [bl or (int(b[r]) < 5) for r, bl in enumerate(black)]
    assert got == exp


def test_black_scan_agg_cpu():
    """Black filter through the fused scan->filter->agg path."""
    rng = np.random.default_rng(11)
    rows = 5000
    a = rng.integers(1, 1000, rows)
    d = rng.integers(0, 11, rows)
    schema, blk = _mk_block([a, d], [abi.ENC_RAW, abi.ENC_DICT])
    import ctypes as C
    offs = (C.c_uint64 * 2)(0, len(blk))
    buf = (C.c_uint8 * len(blk)).from_buffer_copy(bytes(blk))
    bs = abi.BlockSet()
    bs.data = C.cast(buf, C.POINTER(C.c_uint8))
    bs.block_offsets = offs
    bs.n_blocks = 1
    bs.n_cols = 2
    bs.cols = schema
    bs.total_rows = rows
    prog = [abi.BX_COL | 0, abi.BX_COL | 1, abi.BX_MUL,
            abi.BX_CONST | 0, abi.BX_GT]
    filt = abi.make_filter([dict(op=abi.OP_BLACK, bcols=[0, 1],
                                 bprog=prog, bconst=[3000])])
    agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT),
                            dict(kind=abi.AGG_SUM, col_a=0)])
    res = oracle.scan_filter_agg(bs, filt, agg)
    exp_pass = [int(a[r]) * int(d[r]) > 3000 for r in range(rows)]
    assert res.rows_passed == sum(exp_pass)
    got = abi.result_rows(res, 2)
    exp_sum = sum(int(a[r]) for r in range(rows) if exp_pass[r])
    assert got[0][2][1] == exp_sum


def test_black_invalid_programs_rejected():
    rng = np.random.default_rng(0)
    a = rng.integers(0, 100, 64)
    schema, blk = _mk_block([a], [abi.ENC_RAW])
    for bad in ([abi.BX_ADD],                    # underflow
                [abi.BX_COL | 0, abi.BX_COL | 0],  # two results
                [abi.BX_COL | 3],                # col slot out of range
                [0xFF]):                         # unknown opcode
        filt = abi.make_filter([dict(op=abi.OP_BLACK, bcols=[0],
                                     bprog=bad, bconst=[])])
        with pytest.raises(RuntimeError):
            oracle.filter_block(schema, 1, blk, filt)


# ---- GPU parity ---------------------------------------------------------

def _blockset_of(schema, blks, n_cols):
    import ctypes as C
    total = sum(len(b) for b in blks)
    data = np.zeros(total, dtype=np.uint8)
    offs = np.zeros(len(blks) + 1, dtype=np.uint64)
    pos = 0
    for i, b in enumerate(blks):
        offs[i] = pos
        data[pos:pos + len(b)] = np.frombuffer(bytes(b), dtype=np.uint8)
        pos += len(b)
    offs[len(blks)] = pos
    bs = abi.BlockSet()
    bs.data = data.ctypes.data_as(C.POINTER(C.c_uint8))
    bs.block_offsets = offs.ctypes.data_as(C.POINTER(C.c_uint64))
    bs.n_blocks = len(blks)
    bs.n_cols = n_cols
    bs.cols = schema
    bs.total_rows = 0
    bs._keep = (data, offs, schema)
    return bs


@pytest.mark.gpu
@pytest.mark.parametrize("case", range(len(CASES)))
def test_black_filter_gpu_parity(case):
    from oceanbase_amd.engine import GpuEngine
    c = CASES[case]
    rng = np.random.default_rng(77 + case)
    blks = []
    rows = 1400
    schema = None
    avals, bvals = [], []
    for _ in range(3):
        a = rng.choice(np.array([3, 11, 999, 1500, -40], dtype=np.int64),
                       rows)
        b = rng.integers(0, 8, rows) + 1000
        nulls = {0: set(rng.choice(rows, 25, replace=False).tolist())}
        schema, blk = _mk_block([a, b], [abi.ENC_DICT, abi.ENC_RAW], nulls)
        blks.append(blk)
        avals.append(a)
        bvals.append(b)
    bs = _blockset_of(schema, blks, 2)
    filt = abi.make_filter([dict(op=abi.OP_BLACK, bcols=c["bcols"],
                                 bprog=c["prog"], bconst=c["consts"])])
    eng = GpuEngine(0)
    h = eng.load(bs)
    got = eng.filter(h, filt)
    exp = 0
    for i, blk in enumerate(blks):
        _bits, pc = oracle.filter_block(schema, 2, blk, filt)
        exp += pc
    assert got == exp, f"case {case}"
    eng.close()


@pytest.mark.gpu
def test_black_on_dict_lowered_and_agg_gpu():
    """Single-dict-column black program: k_lower_leaves evaluates it once
    per entry into a ref mask (the filter-on-dict optimization); results
    must match the oracle, and the whole plan stays JIT-inlinable."""
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(23)
    rows = 2000
    blks = []
    schema = None
    for _ in range(4):
        d = rng.choice(np.array([0, 2, 5, 7, 10], dtype=np.int64), rows)
        e = rng.integers(10**5, 10**7, rows)
        schema, blk = _mk_block([d, e], [abi.ENC_DICT, abi.ENC_RAW])
        blks.append(blk)
    bs = _blockset_of(schema, blks, 2)
    # (d * d + 1) > 26  -> true for d in {7, 10} only
    prog = [abi.BX_COL | 0, abi.BX_COL | 0, abi.BX_MUL,
            abi.BX_CONST | 0, abi.BX_ADD, abi.BX_CONST | 1, abi.BX_GT]
    filt = abi.make_filter([dict(op=abi.OP_BLACK, bcols=[0],
                                 bprog=prog, bconst=[1, 26])])
    agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT),
                            dict(kind=abi.AGG_SUM, col_a=1)])
    eng = GpuEngine(0)
    h = eng.load(bs)
    res = eng.scan_filter_agg(h, filt, agg)
    ores = oracle.scan_filter_agg(bs, filt, agg)
    assert res.rows_passed == ores.rows_passed
    assert sorted(abi.result_rows(res, 2)) == sorted(
        abi.result_rows(ores, 2))
    eng.close()
