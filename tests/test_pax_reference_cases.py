"""Recreation of the reference's PAX decoder pushdown-filter unit
tests (unittest/storage/blocksstable/encoding/test_const_decoder.cpp,
ROW_CNT=64 from test_column_decoder.h:71): a CONST-encoded column with
exception rows and a trailing null, every white op, with the exact
popcounts the reference asserts — evaluated through the oracle's
scan_filter_agg (the pushdown path this repo replaces)."""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oceanbase_amd import abi, oracle  # noqa: E402

from test_cs_pipeline_equiv import _pax_blockset  # noqa: E402

ROW_CNT = 64


def _const_block(seed_1=0x1, seed_2=0x2, enc=abi.ENC_CONST):
    """61 rows of seed_1, 2 of seed_2, 1 null — the reference's layout:
    0 .. ROW_CNT-3 seed1 | ROW_CNT-3..ROW_CNT-1 seed2 | ROW_CNT-1 null."""
    vals = np.array([seed_1] * (ROW_CNT - 3) + [seed_2] * 2 + [0],
                    dtype=np.int64)
    nb = np.zeros((ROW_CNT + 7) // 8, dtype=np.uint8)
    nb[(ROW_CNT - 1) >> 3] |= 1 << ((ROW_CNT - 1) & 7)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    block = oracle.encode_block(schema, [vals], [enc], [nb])
    return _pax_blockset(schema, [block])


def _count(bs, op, lo=0, hi=0, in_list=None):
    leaf = dict(col=0, op=op, lo=lo, hi=hi)
    if in_list is not None:
        leaf["in_list"] = in_list
    filt = abi.make_filter([leaf])
    res = oracle.scan_filter_agg(bs, filt,
                                 abi.make_agg([], [dict(kind=abi.AGG_COUNT)]))
    return res.rows_passed


@pytest.mark.parametrize("enc", [abi.ENC_CONST, abi.ENC_DICT, abi.ENC_RLE])
def test_filter_push_down_nu_nn_eq_ne(enc):
    """TestConstDecoder::filter_push_down_nu_nn_eq_ne popcounts; also
    run under DICT/RLE (the same shape exercises their null paths)."""
    s1, s2 = 0x1, 0x2
    bs = _const_block(s1, s2, enc)
    assert _count(bs, abi.OP_EQ, s1) == ROW_CNT - 3   # seed1_count
    assert _count(bs, abi.OP_NE, s1) == 2             # seed2 only, null out
    assert _count(bs, abi.OP_EQ, s2) == 2
    assert _count(bs, abi.OP_NE, s2) == ROW_CNT - 3
    assert _count(bs, abi.OP_NU) == 1
    assert _count(bs, abi.OP_NN) == ROW_CNT - 1


@pytest.mark.parametrize("enc", [abi.ENC_CONST, abi.ENC_DICT, abi.ENC_RLE])
def test_filter_push_down_gt_lt_ge_le(enc):
    """TestConstDecoder::filter_push_down_gt_lt_ge_le: refs below,
    between and above the two stored values; nulls never pass."""
    s1, s2 = 10, 20
    bs = _const_block(s1, s2, enc)
    assert _count(bs, abi.OP_GT, s1) == 2
    assert _count(bs, abi.OP_GE, s1) == ROW_CNT - 1
    assert _count(bs, abi.OP_LT, s2) == ROW_CNT - 3
    assert _count(bs, abi.OP_LE, s2) == ROW_CNT - 1
    assert _count(bs, abi.OP_GT, s2) == 0
    assert _count(bs, abi.OP_LT, s1) == 0
    assert _count(bs, abi.OP_GE, 15) == 2
    assert _count(bs, abi.OP_LE, 15) == ROW_CNT - 3


@pytest.mark.parametrize("enc", [abi.ENC_CONST, abi.ENC_DICT, abi.ENC_RLE])
def test_filter_push_down_bt_in(enc):
    """TestConstDecoder::filter_push_down_bt / _in."""
    s1, s2 = 10, 20
    bs = _const_block(s1, s2, enc)
    assert _count(bs, abi.OP_BT, lo=s1, hi=s2) == ROW_CNT - 1
    assert _count(bs, abi.OP_BT, lo=s1 + 1, hi=s2) == 2
    assert _count(bs, abi.OP_BT, lo=s1, hi=s2 - 1) == ROW_CNT - 3
    assert _count(bs, abi.OP_BT, lo=s2 + 1, hi=s2 + 9) == 0
    assert _count(bs, abi.OP_IN, in_list=[s1]) == ROW_CNT - 3
    assert _count(bs, abi.OP_IN, in_list=[s2, 99]) == 2
    assert _count(bs, abi.OP_IN, in_list=[s1, s2]) == ROW_CNT - 1
    assert _count(bs, abi.OP_IN, in_list=[77]) == 0


def test_no_exception_nu_nn():
    """TestConstDecoder::no_exception_nu_nn: a pure const column (no
    exceptions, no nulls)."""
    vals = np.full(ROW_CNT, 0x1, dtype=np.int64)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    bs = _pax_blockset(schema, [oracle.encode_block(
        schema, [vals], [abi.ENC_CONST], None)])
    assert _count(bs, abi.OP_NU) == 0
    assert _count(bs, abi.OP_NN) == ROW_CNT
    assert _count(bs, abi.OP_EQ, 0x1) == ROW_CNT
    assert _count(bs, abi.OP_NE, 0x1) == 0


@pytest.mark.gpu
@pytest.mark.parametrize("enc", [abi.ENC_CONST, abi.ENC_DICT, abi.ENC_RLE])
def test_filter_push_down_gpu_parity(enc):
    """the same reference shape through the GPU filter kernels."""
    from oceanbase_amd.engine import GpuEngine
    s1, s2 = 10, 20
    bs = _const_block(s1, s2, enc)
    eng = GpuEngine(0)
    h = eng.load(bs)
    for op, lo, hi, il, expect in [
            (abi.OP_EQ, s1, 0, None, ROW_CNT - 3),
            (abi.OP_NE, s1, 0, None, 2),
            (abi.OP_NU, 0, 0, None, 1),
            (abi.OP_NN, 0, 0, None, ROW_CNT - 1),
            (abi.OP_GE, 15, 0, None, 2),
            (abi.OP_BT, s1, s2 - 1, None, ROW_CNT - 3),
            (abi.OP_IN, 0, 0, [s2, 99], 2)]:
        leaf = dict(col=0, op=op, lo=lo, hi=hi)
        if il is not None:
            leaf["in_list"] = il
        assert eng.filter(h, abi.make_filter([leaf])) == expect, op
    eng.close()


import pymodel  # noqa: E402


@pytest.mark.parametrize("seed", range(6))
def test_const_with_exceptions_oracle_vs_pymodel(seed):
    """randomized const-dominant columns (value-const or null-const,
    <=255 exceptions incl. nulls): the oracle's encoder output must
    decode identically through the independent pure-Python format
    model."""
    rng = np.random.default_rng(900 + seed)
    rows = int(rng.integers(300, 4000))
    const_val = int(rng.integers(-10**9, 10**9))
    vals = np.full(rows, const_val, dtype=np.int64)
    n_exc = int(rng.integers(1, min(200, rows // 3)))
    exc_rows = sorted(int(x) for x in rng.choice(rows, n_exc,
                                                 replace=False))
    for r in exc_rows:
        vals[r] = const_val + int(rng.integers(-1000, 1000))
    null_rows = set(int(r) for r in exc_rows[:n_exc // 3])
    nb = np.zeros((rows + 7) // 8, dtype=np.uint8)
    for r in null_rows:
        nb[r >> 3] |= 1 << (r & 7)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    block = oracle.encode_block(schema, [vals], [abi.ENC_CONST], [nb])
    got = pymodel.Block(block, [(abi.T_INT, 0, 19, 8)]).decode_col(0)
    for r in range(rows):
        if r in null_rows:
            assert got[r] is None, r
        else:
            assert got[r] == int(vals[r]), r
    # and the oracle pipeline agrees on a count
    bs = _pax_blockset(schema, [block])
    expect = sum(1 for r in range(rows)
                 if r not in null_rows and int(vals[r]) > const_val)
    assert _count(bs, abi.OP_GT, const_val) == expect


def test_const_null_const_with_value_exceptions():
    """null as the most frequent value: const_ref == dict count, real
    values are the exceptions (ob_const_encoder.cpp:79-86)."""
    rows = 500
    vals = np.zeros(rows, dtype=np.int64)
    nb = np.zeros((rows + 7) // 8, dtype=np.uint8)
    exc = {3: 7, 99: -5, 400: 12345}
    for r in range(rows):
        if r in exc:
            vals[r] = exc[r]
        else:
            nb[r >> 3] |= 1 << (r & 7)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    block = oracle.encode_block(schema, [vals], [abi.ENC_CONST], [nb])
    got = pymodel.Block(block, [(abi.T_INT, 0, 19, 8)]).decode_col(0)
    for r in range(rows):
        assert got[r] == exc.get(r), r
    bs = _pax_blockset(schema, [block])
    assert _count(bs, abi.OP_NN) == 3
    assert _count(bs, abi.OP_NU) == rows - 3
    assert _count(bs, abi.OP_GT, 0) == 2
