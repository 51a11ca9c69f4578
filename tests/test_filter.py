"""Pushdown white-filter semantics vs a naive row-wise evaluator.

Mirrors the reference's basic_filter_pushdown_* tests
(unittest/storage/blocksstable/encoding/test_raw_decoder.cpp,
test_const_decoder.cpp: independently computed expected bitmap popcounts per
operator, over every encoder and with NULLs).
"""
import numpy as np
import pytest

from oceanbase_amd import abi, oracle
import pymodel

RNG = np.random.default_rng(7)


def _mk(vals, dtype):
    return np.asarray(vals, dtype=dtype).view(np.uint8)


def _naive_bitmap(vals, leaves, schema_tuples):
    rows = None
    out = []
    for r in range(len(vals[0])):
        ok = True
        for lf in leaves:
            c = lf["col"]
            sc = pymodel.store_class(schema_tuples[c][0])
            ok = ok and pymodel.eval_leaf(lf["op"], vals[c][r],
                                          lf.get("lo", 0), lf.get("hi", 0),
                                          lf.get("in_list", []), sc,
                                          schema_tuples[c][3])
            if not ok:
                break
        out.append(ok)
    return out


def _check(schema_tuples, arrays, encodings, leaves, nulls=None):
    schema = oracle.make_schema(schema_tuples)
    block = oracle.encode_block(schema, list(arrays), encodings, nulls)
    fd = abi.make_filter(leaves)
    bits, popcnt = oracle.filter_block(schema, len(schema_tuples), block, fd)
    pb = pymodel.Block(block, schema_tuples)
    vals = [pb.decode_col(c) for c in range(len(schema_tuples))]
    expect = _naive_bitmap(vals, leaves, schema_tuples)
    assert popcnt == sum(expect)
    for r, e in enumerate(expect):
        got = (bits[r >> 3] >> (r & 7)) & 1
        assert bool(got) == bool(e), (r, vals[0][r] if vals else None)
    return popcnt


ALL_OPS = [abi.OP_EQ, abi.OP_LE, abi.OP_LT, abi.OP_GE, abi.OP_GT, abi.OP_NE]


@pytest.mark.parametrize("enc", [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_RLE,
                                 abi.ENC_INT_DIFF])
@pytest.mark.parametrize("op", ALL_OPS)
def test_int_cmp_ops(enc, op):
    if enc == abi.ENC_RLE:
        vals = np.repeat(RNG.integers(0, 50, 20, dtype=np.int64), 25)
    else:
        vals = RNG.integers(0, 50, 500, dtype=np.int64)
    _check([(abi.T_INT, 0, 19, 8)], [_mk(vals, np.int64)], [enc],
           [dict(col=0, op=op, lo=24)])


@pytest.mark.parametrize("op", ALL_OPS + [abi.OP_BT])
def test_decimal_raw_ops(op):
    vals = RNG.integers(-1000, 1000, 400, dtype=np.int64)
    _check([(abi.T_DECIMAL_INT, 2, 15, 8)], [_mk(vals, np.int64)],
           [abi.ENC_RAW], [dict(col=0, op=op, lo=-100, hi=500)])


def test_bt_and_in():
    vals = RNG.integers(0, 11, 600, dtype=np.int64)
    _check([(abi.T_DECIMAL_INT, 2, 15, 8)], [_mk(vals, np.int64)],
           [abi.ENC_DICT], [dict(col=0, op=abi.OP_BT, lo=5, hi=7)])
    _check([(abi.T_DECIMAL_INT, 2, 15, 8)], [_mk(vals, np.int64)],
           [abi.ENC_DICT], [dict(col=0, op=abi.OP_IN, in_list=[1, 4, 9])])


def test_null_semantics():
    rows = 300
    vals = RNG.integers(0, 100, rows, dtype=np.int64)
    nb = np.zeros((rows + 7) // 8, dtype=np.uint8)
    null_rows = list(range(0, rows, 7))
    for r in null_rows:
        nb[r >> 3] |= 1 << (r & 7)
    schema_t = [(abi.T_INT, 0, 19, 8)]
    # comparisons never match NULL; NU matches exactly the null rows
    pc_lt = _check(schema_t, [_mk(vals, np.int64)], [abi.ENC_RAW],
                   [dict(col=0, op=abi.OP_LT, lo=200)], [nb])
    assert pc_lt == rows - len(null_rows)
    pc_nu = _check(schema_t, [_mk(vals, np.int64)], [abi.ENC_RAW],
                   [dict(col=0, op=abi.OP_NU)], [nb])
    assert pc_nu == len(null_rows)
    pc_nn = _check(schema_t, [_mk(vals, np.int64)], [abi.ENC_RAW],
                   [dict(col=0, op=abi.OP_NN)], [nb])
    assert pc_nn == rows - len(null_rows)


def test_char_cmp():
    vals = RNG.choice(np.frombuffer(b"ANR", dtype=np.uint8), 500)
    for op in ALL_OPS:
        _check([(abi.T_CHAR, 0, 0, 1)], [vals], [abi.ENC_DICT],
               [dict(col=0, op=op, lo=ord("N"))])


def test_date_range_and():
    rows = 800
    ship = RNG.integers(8036, 10600, rows, dtype=np.int32)
    disc = RNG.integers(0, 11, rows, dtype=np.int64)
    qty = RNG.integers(100, 5100, rows, dtype=np.int64)
    d94 = oracle.date_days(1994, 1, 1)
    d95 = oracle.date_days(1995, 1, 1)
    # Q6 predicate shape
    _check([(abi.T_DATE, 0, 0, 4), (abi.T_DECIMAL_INT, 2, 15, 8),
            (abi.T_DECIMAL_INT, 2, 15, 8)],
           [_mk(ship, np.int32), _mk(disc, np.int64), _mk(qty, np.int64)],
           [abi.ENC_INT_DIFF, abi.ENC_DICT, abi.ENC_DICT],
           [dict(col=0, op=abi.OP_GE, lo=d94),
            dict(col=0, op=abi.OP_LT, lo=d95),
            dict(col=1, op=abi.OP_BT, lo=5, hi=7),
            dict(col=2, op=abi.OP_LT, lo=2400)])


def test_const_filter():
    vals = np.full(200, 42, dtype=np.int64)
    assert _check([(abi.T_INT, 0, 19, 8)], [_mk(vals, np.int64)],
                  [abi.ENC_CONST], [dict(col=0, op=abi.OP_EQ, lo=42)]) == 200
    assert _check([(abi.T_INT, 0, 19, 8)], [_mk(vals, np.int64)],
                  [abi.ENC_CONST], [dict(col=0, op=abi.OP_LT, lo=42)]) == 0


def test_no_filter_passes_all():
    vals = RNG.integers(0, 100, 100, dtype=np.int64)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    blk = oracle.encode_block(schema, [_mk(vals, np.int64)], [abi.ENC_RAW])
    fd = abi.FilterDesc()
    fd.n_leaves = 0
    bits, pc = oracle.filter_block(schema, 1, blk, fd)
    assert pc == 100
