"""The reference's shared pushdown-filter matrix
(TestColumnDecoder::basic_filter_pushdown_* in
unittest/storage/blocksstable/encoding/test_column_decoder.h:696-1050,
instantiated per decoder in test_general_column_decoder.cpp): ROW_CNT=64
rows split 24/10/10/10 across four ascending seeds plus 10 trailing
nulls, every white op with the popcounts the reference asserts — here
run across EVERY obx PAX encoding of the column (the reference
instantiates the same matrix for DICT/RLE/INTDIFF/HEX/STRING_DIFF/
STRING_PREFIX/COLUMN_EQUAL/COLUMN_SUBSTR)."""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oceanbase_amd import abi, oracle  # noqa: E402

from test_cs_pipeline_equiv import _pax_blockset  # noqa: E402

ROW_CNT = 64
SEG = [24, 10, 10, 10, 10]  # seed0 x24, seed1/2/3 x10, null x10


def _counts(op, lo_i, hi_i=None, in_i=None):
    """expected popcount over the seed segments (segment k holds seed k;
    the last segment is null and fails every value op)."""
    n = 0
    for k, cnt in enumerate(SEG[:4]):
        if op == abi.OP_EQ:
            ok = k == lo_i
        elif op == abi.OP_NE:
            ok = k != lo_i
        elif op == abi.OP_LT:
            ok = k < lo_i
        elif op == abi.OP_LE:
            ok = k <= lo_i
        elif op == abi.OP_GT:
            ok = k > lo_i
        elif op == abi.OP_GE:
            ok = k >= lo_i
        elif op == abi.OP_BT:
            ok = lo_i <= k <= hi_i
        elif op == abi.OP_IN:
            ok = k in in_i
        else:
            ok = False
        if ok:
            n += cnt
    if op == abi.OP_NU:
        n = SEG[4]
    if op == abi.OP_NN:
        n = ROW_CNT - SEG[4]
    return n


def _nulls_tail():
    nb = np.zeros((ROW_CNT + 7) // 8, dtype=np.uint8)
    for r in range(ROW_CNT - 10, ROW_CNT):
        nb[r >> 3] |= 1 << (r & 7)
    return nb


def _int_bs(enc, seeds):
    vals = np.zeros(ROW_CNT, dtype=np.int64)
    p = 0
    for k, cnt in enumerate(SEG[:4]):
        vals[p:p + cnt] = seeds[k]
        p += cnt
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    block = oracle.encode_block(schema, [vals], [enc], [_nulls_tail()])
    return _pax_blockset(schema, [block])


def _char_bs(enc, seeds, ln=4, extra_coleq=False):
    data = np.zeros(ROW_CNT * ln, dtype=np.uint8)
    p = 0
    for k, cnt in enumerate(SEG[:4]):
        s = seeds[k]
        for r in range(p, p + cnt):
            data[r * ln:(r + 1) * ln] = np.frombuffer(s, dtype=np.uint8)
        p += cnt
    if not extra_coleq:
        schema = oracle.make_schema([(abi.T_CHAR, 0, 0, ln)])
        block = oracle.encode_block(schema, [data], [enc], [_nulls_tail()])
        return _pax_blockset(schema, [block]), 0
    # span encodings need a reference column: col0 RAW carries the same
    # values, col1 is the span-encoded copy the matrix filters on
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, ln)] * 2)
    block = oracle.encode_block(schema, [data.copy(), data.copy()],
                                [abi.ENC_RAW, enc],
                                [_nulls_tail(), _nulls_tail()])
    return _pax_blockset(schema, [block]), 1


def _run_matrix(bs, col, keys):
    """keys: the four seed values as leaf operands (int64)."""
    def count(op, lo=0, hi=0, il=None):
        leaf = dict(col=col, op=op, lo=lo, hi=hi)
        if il is not None:
            leaf["in_list"] = il
        res = oracle.scan_filter_agg(
            bs, abi.make_filter([leaf]),
            abi.make_agg([], [dict(kind=abi.AGG_COUNT)]))
        return res.rows_passed

    s5 = keys[3] + 12345  # a value not present (seed5 analogue)
    assert count(abi.OP_EQ, keys[0]) == _counts(abi.OP_EQ, 0)
    assert count(abi.OP_EQ, keys[1]) == _counts(abi.OP_EQ, 1)
    assert count(abi.OP_NE, keys[0]) == _counts(abi.OP_NE, 0)
    assert count(abi.OP_NU) == _counts(abi.OP_NU, 0)
    assert count(abi.OP_NN) == _counts(abi.OP_NN, 0)
    assert count(abi.OP_LT, keys[1]) == _counts(abi.OP_LT, 1)
    assert count(abi.OP_LE, keys[1]) == _counts(abi.OP_LE, 1)
    assert count(abi.OP_GT, keys[1]) == _counts(abi.OP_GT, 1)
    assert count(abi.OP_GE, keys[1]) == _counts(abi.OP_GE, 1)
    assert count(abi.OP_BT, keys[1], keys[2]) == _counts(abi.OP_BT, 1, 2)
    assert count(abi.OP_BT, keys[0], keys[3]) == _counts(abi.OP_BT, 0, 3)
    assert count(abi.OP_IN, il=[keys[1], keys[2], s5]) == \
        _counts(abi.OP_IN, 0, in_i={1, 2})
    assert count(abi.OP_IN, il=[s5]) == 0


INT_SEEDS = [10000, 10001, 10002, 10003]


@pytest.mark.parametrize("enc", [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_RLE,
                                 abi.ENC_INT_DIFF, abi.ENC_CONST])
def test_int_matrix(enc):
    bs = _int_bs(enc, INT_SEEDS)
    _run_matrix(bs, 0, INT_SEEDS)


CHAR_SEEDS = [b"s100", b"s101", b"s102", b"s103"]
CHAR_KEYS = [int.from_bytes(s, "little") for s in CHAR_SEEDS]


@pytest.mark.parametrize("enc", [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_RLE,
                                 abi.ENC_HEX, abi.ENC_SDIFF,
                                 abi.ENC_STRING_PREFIX])
def test_char_matrix(enc):
    bs, col = _char_bs(enc, CHAR_SEEDS)
    _run_matrix(bs, col, CHAR_KEYS)


@pytest.mark.parametrize("enc", [abi.ENC_COLUMN_EQUAL,
                                 abi.ENC_COLUMN_SUBSTR])
def test_span_matrix(enc):
    """TestColumnEqualMicroDecoder / TestInterColumnSubstringDecoder
    instantiations: the matrix on a span-encoded column."""
    bs, col = _char_bs(enc, CHAR_SEEDS, extra_coleq=True)
    _run_matrix(bs, col, CHAR_KEYS)


# ---- the same matrix over CS-format columns (recreating the shapes of
# unittest/storage/blocksstable/cs_encoding/test_integer_pd_filter.cpp,
# test_int_dict_pd_filter.cpp, test_str_dict_pd_filter.cpp,
# test_string_pd_filter.cpp through the transformer-role load path) ----

import cs_oracle_util as cs  # noqa: E402
from test_cs_block import (  # noqa: E402
    _enc as cs_enc, _int_col as cs_int_col, _str_col as cs_str_col,
)


def _cs_int_bs(seeds, dict_=False, enc=0):
    vals, p = [0] * ROW_CNT, 0
    for k, cnt in enumerate(SEG[:4]):
        for r in range(p, p + cnt):
            vals[r] = seeds[k]
        p += cnt
    nulls = list(range(ROW_CNT - 10, ROW_CNT))
    block = cs_enc(ROW_CNT, [cs_int_col(vals, null_rows=nulls,
                                        dict_=dict_, enc=enc)])
    schema, pax = cs.to_pax_blocks([block])
    return _pax_blockset(schema, pax)


def _cs_str_bs(seeds, dict_=False):
    strs, p = [b""] * ROW_CNT, 0
    for k, cnt in enumerate(SEG[:4]):
        for r in range(p, p + cnt):
            strs[r] = seeds[k]
        p += cnt
    nulls = list(range(ROW_CNT - 10, ROW_CNT))
    for r in nulls:
        strs[r] = seeds[0]  # placeholder; bytes skipped for null rows
    block = cs_enc(ROW_CNT, [cs_str_col(strs, null_rows=nulls,
                                        dict_=dict_)])
    schema, pax = cs.to_pax_blocks([block])
    return _pax_blockset(schema, pax)


@pytest.mark.parametrize("dict_,enc", [(False, 0), (False, 6), (True, 0)])
def test_cs_integer_matrix(dict_, enc):
    """test_integer_pd_filter / test_int_dict_pd_filter shapes."""
    bs = _cs_int_bs(INT_SEEDS, dict_=dict_, enc=enc)
    _run_matrix(bs, 0, INT_SEEDS)


@pytest.mark.parametrize("dict_", [False, True])
def test_cs_string_matrix(dict_):
    """test_string_pd_filter / test_str_dict_pd_filter shapes (fixed-len
    strings; the load path maps them to char(N) PAX columns)."""
    bs = _cs_str_bs(CHAR_SEEDS, dict_=dict_)
    _run_matrix(bs, 0, CHAR_KEYS)


def test_exceed_range_compare_filter():
    """TestIntegerPdFilter::test_exceed_range_compare_filter: operands
    outside the column's stored domain must degenerate to all/none, not
    wrap (the packed-domain lowering biases signed domains)."""
    vals = np.array([100, 200, 300, 150, 250] * 12 + [175, 225, 275, 125],
                    dtype=np.int64)
    rows = len(vals)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    for enc in (abi.ENC_RAW, abi.ENC_DICT, abi.ENC_INT_DIFF):
        bs = _pax_blockset(schema, [oracle.encode_block(
            schema, [vals.copy()], [enc], None)])
        def cnt(op, lo=0, hi=0):
            return oracle.scan_filter_agg(
                bs, abi.make_filter([dict(col=0, op=op, lo=lo, hi=hi)]),
                abi.make_agg([], [dict(kind=abi.AGG_COUNT)])).rows_passed
        assert cnt(abi.OP_GT, 2**63 - 1) == 0
        assert cnt(abi.OP_LT, -2**63) == 0
        assert cnt(abi.OP_GE, -2**63) == rows
        assert cnt(abi.OP_LE, 2**63 - 1) == rows
        assert cnt(abi.OP_BT, -2**63, 2**63 - 1) == rows
        assert cnt(abi.OP_BT, 10**15, 10**16) == 0
        assert cnt(abi.OP_EQ, 10**12) == 0
        assert cnt(abi.OP_NE, 10**12) == rows


def test_all_null_integer_through_cs_load():
    """TestIntegerPdFilter::test_all_null_integer_decoder through the
    CS load path: an all-null CS integer column must stay all-null
    after the transform and fail every value filter."""
    rows = 200
    block = cs_enc(rows, [cs_int_col([0] * rows,
                                     null_rows=list(range(rows)))])
    schema, pax = cs.to_pax_blocks([block])
    bs = _pax_blockset(schema, pax)
    def cnt(op, lo=0):
        return oracle.scan_filter_agg(
            bs, abi.make_filter([dict(col=0, op=op, lo=lo)]),
            abi.make_agg([], [dict(kind=abi.AGG_COUNT)])).rows_passed
    assert cnt(abi.OP_NU) == rows
    assert cnt(abi.OP_NN) == 0
    assert cnt(abi.OP_EQ, 0) == 0
    assert cnt(abi.OP_GE, -2**62) == 0


def test_integer_in_at_capacity():
    """TestIntegerPdFilter::test_integer_many_in_values analogue at the
    engine's OBX_MAX_IN_LIST=8 bound (larger lists are the reference
    adapter's fallback; include/obx.h:149)."""
    vals = np.arange(64, dtype=np.int64) % 16
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    bs = _pax_blockset(schema, [oracle.encode_block(
        schema, [vals], [abi.ENC_RAW], None)])
    il = [0, 2, 4, 6, 8, 10, 12, 14]  # exactly 8 entries
    res = oracle.scan_filter_agg(
        bs, abi.make_filter([dict(col=0, op=abi.OP_IN, in_list=il)]),
        abi.make_agg([], [dict(kind=abi.AGG_COUNT)]))
    assert res.rows_passed == sum(1 for v in vals if int(v) in set(il))
