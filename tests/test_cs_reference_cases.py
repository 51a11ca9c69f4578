"""Recreation of the reference's own integer-stream unit-test matrix
(unittest/storage/blocksstable/cs_encoding/test_integer_stream.cpp):
each named case generates datums of the same shape (spans, monotonic
patterns, null modes) and checks, across EVERY implemented stream
encoding type, that the oracle's encoder picks the meta the reference's
harness expects (USE_BASE / REPLACE_NULL attributes, UintWidth tag) and
that decode is bit-exact with nulls recovered the same way."""
import ctypes as C
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oceanbase_amd import oracle  # noqa: E402

_lib = C.CDLL(os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "oracle", "liboracle.so"))


class IntMeta(C.Structure):
    _fields_ = [("version", C.c_uint8), ("attr", C.c_uint8),
                ("type", C.c_uint8), ("width_tag", C.c_uint8),
                ("base", C.c_uint64), ("null_replaced", C.c_uint64),
                ("precision_width_tag", C.c_uint8),
                ("pfor_packing_type", C.c_uint8)]


_lib.obx_cs_int_stream_enc3.restype = C.c_int64
_lib.obx_cs_int_stream_enc3.argtypes = [
    C.POINTER(C.c_int64), C.POINTER(C.c_uint8), C.c_uint32, C.c_uint8,
    C.c_int, C.c_int64, C.POINTER(C.c_uint8), C.c_size_t]
_lib.obx_cs_int_stream_dec.restype = C.c_int64
_lib.obx_cs_int_stream_dec.argtypes = [
    C.POINTER(C.c_uint8), C.c_size_t, C.c_uint32, C.POINTER(C.c_int64),
    C.POINTER(IntMeta)]

USE_BASE, REPLACE_NULL = 0x1, 0x2
ALL_TYPES = [1, 2, 3, 4, 5, 6, 8]


def _run(vals, nulls=None, enc_type=1, replace=None):
    rows = len(vals)
    v = np.asarray(vals, dtype=np.int64)
    nb = None
    nbp = None
    if nulls:
        nb = np.zeros((rows + 7) // 8, dtype=np.uint8)
        for r in nulls:
            nb[r >> 3] |= 1 << (r & 7)
        nbp = nb.ctypes.data_as(C.POINTER(C.c_uint8))
    cap = 64 + rows * 24
    buf = (C.c_uint8 * cap)()
    n = _lib.obx_cs_int_stream_enc3(
        v.ctypes.data_as(C.POINTER(C.c_int64)), nbp, rows, enc_type,
        1 if replace is not None else 0,
        replace if replace is not None else 0, buf, cap)
    assert n > 0, (enc_type, len(vals))
    out = np.zeros(rows, dtype=np.int64)
    m = IntMeta()
    assert _lib.obx_cs_int_stream_dec(
        buf, n, rows, out.ctypes.data_as(C.POINTER(C.c_int64)),
        C.byref(m)) == n
    return out, m, n


def _check_roundtrip(vals, nulls, out, m, replace):
    nulls = set(nulls or [])
    for r, x in enumerate(vals):
        if r in nulls:
            if replace is not None:
                assert m.attr & REPLACE_NULL
                assert out[r] == replace  # recovered by equality
        else:
            assert int(out[r]) == x, r


def _shape(kind, rows, lo, hi, rng):
    if kind == "random":
        return [int(x) for x in rng.integers(lo, hi + 1, rows,
                                             dtype=np.int64)]
    if kind == "inc":
        return [lo + i for i in range(rows)]
    if kind == "dec":
        return [hi - i for i in range(rows)]
    if kind == "equal":
        return [lo] * rows
    raise AssertionError(kind)


@pytest.mark.parametrize("enc_type", ALL_TYPES)
def test_all_zero(enc_type):
    """TEST_F test_all_zero: min=max=0 in the three null modes; the
    chosen width is 1 byte and no base is stored (min >= 0)."""
    rows = 300
    # j=0: no null
    out, m, _ = _run([0] * rows, enc_type=enc_type)
    assert not (m.attr & USE_BASE) and m.width_tag == 0
    assert list(out) == [0] * rows
    # j=1: null via bitmap (no replace): null slots decode to 0
    nulls = list(range(0, rows, 7))
    out, m, _ = _run([0] * rows, nulls=nulls, enc_type=enc_type)
    assert not (m.attr & REPLACE_NULL)
    assert list(out) == [0] * rows
    # j=2: REPLACE_NULL_VALUE with replace value 2 (the harness's pick)
    out, m, _ = _run([0] * rows, nulls=nulls, enc_type=enc_type,
                     replace=2)
    assert m.attr & REPLACE_NULL and m.null_replaced == 2
    _check_roundtrip([0] * rows, nulls, out, m, 2)


@pytest.mark.parametrize("enc_type", ALL_TYPES)
@pytest.mark.parametrize("span,wtag", [
    ((-2**7, 2**7 - 1), 0), ((-2**15, 2**15 - 1), 1),
    ((-2**31, 2**31 - 1), 2), ((-2**63, 2**63 - 1), 3)])
def test_all_min_max(enc_type, span, wtag):
    """TEST_F test_all_min_max: full intN spans; USE_BASE expected
    (negative minimum) and the width tag covers the range."""
    lo, hi = span
    rng = np.random.default_rng(3 + wtag)
    vals = _shape("random", 400, lo, hi, rng)
    vals[0], vals[1] = lo, hi  # pin the extremes
    out, m, _ = _run(vals, enc_type=enc_type)
    assert m.attr & USE_BASE
    assert m.base == (lo & ((1 << 64) - 1))
    assert m.width_tag == wtag
    assert [int(x) for x in out] == vals
    # with a null bitmap (j=1): same meta, null slots hold the base
    nulls = [5, 55]
    out, m, _ = _run(vals, nulls=nulls, enc_type=enc_type)
    assert m.attr & USE_BASE and not (m.attr & REPLACE_NULL)
    _check_roundtrip(vals, nulls, out, m, None)


@pytest.mark.parametrize("enc_type", ALL_TYPES)
def test_negative_inc_delta(enc_type):
    """TEST_F test_negative_inc_delta: strict increment from
    INT32_MIN+2 and strict decrement down from UINT32_MAX+1."""
    vals = [(-2**31 + 2) + i for i in range(500)]
    out, m, _ = _run(vals, enc_type=enc_type)
    assert m.attr & USE_BASE
    assert [int(x) for x in out] == vals
    vals = [(2**32 + 1) - i for i in range(500)]
    out, m, _ = _run(vals, enc_type=enc_type)
    assert not (m.attr & USE_BASE)  # all positive
    assert [int(x) for x in out] == vals


@pytest.mark.parametrize("enc_type", ALL_TYPES)
@pytest.mark.parametrize("kind,start", [
    ("inc", 5), ("inc", -2**31), ("inc", -5),
    ("dec", 2**31 - 1), ("dec", 5), ("dec", -5),
    ("equal", 5), ("equal", -5), ("equal", 0)])
def test_delta_shapes(enc_type, kind, start):
    """TEST_F test_delta: monotonic / constant runs from the
    reference's start points, 8-byte datums."""
    rows = 400
    if kind == "inc":
        vals = [start + i for i in range(rows)]
    elif kind == "dec":
        vals = [start - i for i in range(rows)]
    else:
        vals = [start] * rows
    out, m, _ = _run(vals, enc_type=enc_type)
    assert [int(x) for x in out] == vals
    if min(vals) < 0:
        assert m.attr & USE_BASE


@pytest.mark.parametrize("enc_type", ALL_TYPES)
def test_all_0_and_replace_value_negative(enc_type):
    """TEST_F test_all_0_and_replace_value_negative: all-zero data with
    replace value -1 -> USE_BASE (base = -1) and 1-byte width."""
    rows = 200
    nulls = [0, 9, 100]
    out, m, _ = _run([0] * rows, nulls=nulls, enc_type=enc_type,
                     replace=-1)
    assert m.attr & USE_BASE and m.attr & REPLACE_NULL
    assert m.base == ((-1) & ((1 << 64) - 1))
    assert m.width_tag == 0
    _check_roundtrip([0] * rows, nulls, out, m, -1)


@pytest.mark.parametrize("enc_type", ALL_TYPES)
def test_raw_negative_with_null(enc_type):
    """TEST_F test_raw_negative_with_null: data in [-100, 0], replace
    value -101 (min-1, the column rule)."""
    rng = np.random.default_rng(8)
    rows = 300
    vals = [int(x) for x in rng.integers(-100, 1, rows)]
    nulls = [int(x) for x in rng.choice(rows, 20, replace=False)]
    out, m, _ = _run(vals, nulls=nulls, enc_type=enc_type, replace=-101)
    assert m.attr & USE_BASE and m.attr & REPLACE_NULL
    assert m.null_replaced == ((-101) & ((1 << 64) - 1))
    _check_roundtrip(vals, nulls, out, m, -101)


@pytest.mark.parametrize("enc_type", ALL_TYPES)
def test_uint32_encoding(enc_type):
    """TEST_F test_uint32_encoding: unsigned 32-bit span."""
    rng = np.random.default_rng(12)
    vals = [int(x) for x in rng.integers(0, 2**32, 500,
                                         dtype=np.uint64)]
    out, m, _ = _run(vals, enc_type=enc_type)
    assert not (m.attr & USE_BASE)
    assert m.width_tag == 2
    assert [int(x) for x in out] == vals


@pytest.mark.parametrize("enc_type", ALL_TYPES)
def test_rle_shape(enc_type):
    """TEST_F test_rle: long runs of repeated values."""
    rng = np.random.default_rng(21)
    vals = []
    while len(vals) < 600:
        vals += [int(rng.integers(-50, 50))] * int(rng.integers(1, 40))
    vals = vals[:600]
    out, m, _ = _run(vals, enc_type=enc_type)
    assert [int(x) for x in out] == vals


# ---- block/column-level cases recreated from test_cs_encoder.cpp ----

from test_cs_block import (  # noqa: E402
    _enc as blk_enc, _dec as blk_dec, _get_int as blk_get_int,
    _int_col as blk_int_col, CA_HAS_NULL_BITMAP,
)


def _col_meta(v, c):
    """decode the column's stream meta from the block view."""
    cv = v.col[c]
    m = IntMeta()
    out = np.zeros(v.rows, dtype=np.int64)
    n = _lib.obx_cs_int_stream_dec(
        C.cast(cv.int_stream, C.POINTER(C.c_uint8)), cv.int_stream_len,
        v.rows, out.ctypes.data_as(C.POINTER(C.c_int64)), C.byref(m))
    assert n > 0
    return m


def test_cs_encoder_integer_cases():
    """TestCSEncoder::test_integer_encoder cases <1>-<7> on an int32
    column (store_width 4): the exact width/base/replace expectations
    the reference asserts."""
    i32min, i32max = -2**31, 2**31 - 1
    # <1> -50..49: width 1, base -50, no replace
    v = blk_dec(blk_enc(100, [blk_int_col(list(range(-50, 50)),
                                          store_width=4)]))
    m = _col_meta(v, 0)
    assert m.width_tag == 0 and (m.attr & USE_BASE)
    assert m.base == ((-50) & ((1 << 64) - 1))
    assert not (m.attr & REPLACE_NULL)
    # <2> INT32_MIN/NULL/-1: width 4, replace value 0 (min == store min
    # -> largest-not-existed = max+1 = 0), base INT32_MIN
    v = blk_dec(blk_enc(3, [blk_int_col([i32min, 0, -1], null_rows=[1],
                                        store_width=4)]))
    m = _col_meta(v, 0)
    assert m.width_tag == 2 and (m.attr & REPLACE_NULL)
    assert m.null_replaced == 0
    assert m.base == (i32min & ((1 << 64) - 1))
    out, nulls = blk_get_int(v, 0)
    assert nulls == {1} and out[0] == i32min and out[2] == -1
    # <3> full int32 span + null: bitmap (no adjacent value left),
    # width 4, base INT32_MIN
    vals = [i32min, i32max, 0] + list(range(-100, 100))
    v = blk_dec(blk_enc(len(vals), [blk_int_col(vals, null_rows=[2],
                                                store_width=4)]))
    assert v.col[0].attrs & CA_HAS_NULL_BITMAP
    m = _col_meta(v, 0)
    assert not (m.attr & REPLACE_NULL) and m.width_tag == 2
    assert m.base == (i32min & ((1 << 64) - 1))
    out, nulls = blk_get_int(v, 0)
    assert nulls == {2} and out[0] == i32min and out[1] == i32max
    # <4> 0/NULL/INT32_MAX: replace -1 (max == store max, min == 0)
    v = blk_dec(blk_enc(3, [blk_int_col([0, 0, i32max], null_rows=[1],
                                        store_width=4)]))
    m = _col_meta(v, 0)
    assert (m.attr & REPLACE_NULL)
    assert m.null_replaced == ((-1) & ((1 << 64) - 1))
    # <5> 0/NULL/INT32_MAX-1: replace INT32_MAX (max+1)
    v = blk_dec(blk_enc(3, [blk_int_col([0, 0, i32max - 1],
                                        null_rows=[1], store_width=4)]))
    m = _col_meta(v, 0)
    assert (m.attr & REPLACE_NULL) and m.null_replaced == i32max
    # <6> 0..999 monotonic: width 2, no base, no bitmap, no replace
    v = blk_dec(blk_enc(1000, [blk_int_col(list(range(1000)),
                                           store_width=4)]))
    m = _col_meta(v, 0)
    assert m.width_tag == 1
    assert not (m.attr & (USE_BASE | REPLACE_NULL))
    assert not (v.col[0].attrs & CA_HAS_NULL_BITMAP)
    # <7> all null: width 1, replace value 0, no base, no bitmap
    v = blk_dec(blk_enc(1000, [blk_int_col([0] * 1000,
                                           null_rows=list(range(1000)),
                                           store_width=4)]))
    m = _col_meta(v, 0)
    assert m.width_tag == 0 and (m.attr & REPLACE_NULL)
    assert m.null_replaced == 0 and not (m.attr & USE_BASE)
    assert not (v.col[0].attrs & CA_HAS_NULL_BITMAP)
    out, nulls = blk_get_int(v, 0)
    assert nulls == set(range(1000))


def test_cs_encoder_dict_const_ref_cases():
    """TestCSEncoder::test_dict_const_ref_encoder: five columns with 64
    exceptions each probing the const-ref decision from both sides
    (value-const vs null-const vs too-spread)."""
    from test_cs_block import _str_col as blk_str_col
    rows, ec = 1000, 64
    base = rows - ec
    # col0: 936 x 1000 then 0..63 -> const ref (value const), dict 65
    c0 = [1000] * base + list(range(ec))
    # col1: 936 x "a"*10 then 64 x "a"*20 -> str const ref, const ref 0
    c1 = [b"a" * 10] * base + [b"a" * 20] * ec
    # col2: i%2+1000 with 64 trailing nulls -> too spread, no const
    c2 = [i % 2 + 1000 for i in range(base)] + [0] * ec
    c2_nulls = list(range(base, rows))
    # col3: 936 nulls then "a"*100 / "a"*101 -> NULL is the const value
    c3 = [b""] * base + [b"a" * (100 + i % 2) for i in range(ec)]
    c3_nulls = list(range(base))
    # col4: 936 x 1000 then 64 nulls -> const 1000, null exceptions
    c4 = [1000] * base + [0] * ec
    c4_nulls = list(range(base, rows))
    v = blk_dec(blk_enc(rows, [
        blk_int_col(c0, dict_=True),
        blk_str_col(c1, dict_=True),
        blk_int_col(c2, null_rows=c2_nulls, dict_=True),
        blk_str_col(c3, null_rows=c3_nulls, dict_=True),
        blk_int_col(c4, null_rows=c4_nulls, dict_=True),
    ]))
    CONST = 0x4
    # col0: const, ec exceptions, dict of 65
    assert v.col[0].dm_attrs & CONST
    assert v.col[0].dm_ref_row_cnt == 2 + 2 * ec
    assert v.col[0].dm_distinct == 1 + ec
    # col1: const, 2 distinct, the shorter string sorts first (ref 0
    # is the const -- the reference asserts const_node_.dict_ref_ == 0)
    assert v.col[1].dm_attrs & CONST
    assert v.col[1].dm_distinct == 2
    assert v.col[1].dm_ref_row_cnt == 2 + 2 * ec
    # col2: 468/468/64 split -> exceptions over 10% -> plain refs
    assert not (v.col[2].dm_attrs & CONST)
    assert v.col[2].dm_ref_row_cnt == rows
    # col3: null is the const value (the reference asserts max_ref_ == 2
    # and const ref == 2 == distinct_val_cnt)
    assert v.col[3].dm_attrs & CONST
    assert v.col[3].dm_distinct == 2
    assert v.col[3].dm_ref_row_cnt == 2 + 2 * ec
    # col4: const 1000 with null exceptions (max_ref 1, const ref 0)
    assert v.col[4].dm_attrs & CONST
    assert v.col[4].dm_distinct == 1
    assert v.col[4].dm_ref_row_cnt == 2 + 2 * ec
    # roundtrips
    out, nulls = blk_get_int(v, 0)
    assert list(out) == c0 and nulls == set()
    out, nulls = blk_get_int(v, 2)
    assert nulls == set(c2_nulls)
    assert [int(x) for x in out[:base]] == c2[:base]
    out, nulls = blk_get_int(v, 4)
    assert nulls == set(c4_nulls)
    assert all(int(x) == 1000 for x in out[:base])
    from test_cs_block import _get_str as blk_get_str
    srows, snulls = blk_get_str(v, 1)
    assert srows == c1 and snulls == set()
    srows, snulls = blk_get_str(v, 3)
    assert snulls == set(c3_nulls)
    assert srows[base:] == c3[base:]


def test_cs_encoder_string_cases():
    """TestCSEncoder::test_string_encoder cases <1>-<5>: the
    fixed/var/zero-len-null/bitmap decision matrix the reference
    asserts per shape."""
    from test_cs_block import (
        _str_col as blk_str_col, _get_str as blk_get_str,
        CA_IS_FIXED, STR_ZERO_LEN_NULL, STR_FIXED_LEN,
    )
    # <1> 100 fixed-64 strings + 1 null: IS_FIXED + null bitmap, not
    # zero-len-null (padding+bitmap beats an offset stream)
    strs = [bytes([65 + i % 26]) * 64 for i in range(100)] + [b""]
    v = blk_dec(blk_enc(101, [blk_str_col(strs, null_rows=[100])]))
    assert v.col[0].attrs & CA_IS_FIXED
    assert v.col[0].attrs & CA_HAS_NULL_BITMAP
    assert v.col[0].sm_attr & STR_FIXED_LEN
    assert v.col[0].sm_fixed_str_len == 64
    assert not (v.col[0].sm_attr & STR_ZERO_LEN_NULL)
    out, nulls = blk_get_str(v, 0)
    assert nulls == {100} and out[:100] == strs[:100]
    # <2> var lens 1..100 + null, no zero-length datum: var with
    # zero-len-as-null, NO bitmap
    strs = [b"\x0f" * (i + 1) for i in range(100)] + [b""]
    v = blk_dec(blk_enc(101, [blk_str_col(strs, null_rows=[100])]))
    assert not (v.col[0].attrs & CA_IS_FIXED)
    assert not (v.col[0].attrs & CA_HAS_NULL_BITMAP)
    assert v.col[0].sm_attr & STR_ZERO_LEN_NULL
    out, nulls = blk_get_str(v, 0)
    assert nulls == {100} and out[:100] == strs[:100]
    # <3> var lens 0..N + null, WITH a zero-length datum: bitmap, var,
    # not zero-len-null
    strs = [b"\x0f" * i for i in range(60)] + [b""]
    v = blk_dec(blk_enc(61, [blk_str_col(strs, null_rows=[60])]))
    assert not (v.col[0].attrs & CA_IS_FIXED)
    assert v.col[0].attrs & CA_HAS_NULL_BITMAP
    assert not (v.col[0].sm_attr & STR_ZERO_LEN_NULL)
    out, nulls = blk_get_str(v, 0)
    assert nulls == {60}
    assert out[0] == b"" and 0 not in nulls  # real empty vs null
    # <4> all null: var + zero-len-as-null, no bitmap
    v = blk_dec(blk_enc(100, [blk_str_col([b""] * 100,
                                          null_rows=list(range(100)))]))
    assert not (v.col[0].attrs & CA_IS_FIXED)
    assert not (v.col[0].attrs & CA_HAS_NULL_BITMAP)
    assert v.col[0].sm_attr & STR_ZERO_LEN_NULL
    out, nulls = blk_get_str(v, 0)
    assert nulls == set(range(100))
    # <5> all zero-length REAL datums: fixed with fixed_len 0
    v = blk_dec(blk_enc(100, [blk_str_col([b""] * 100)]))
    assert v.col[0].attrs & CA_IS_FIXED
    assert v.col[0].sm_fixed_str_len == 0
    out, nulls = blk_get_str(v, 0)
    assert nulls == set() and all(s == b"" for s in out)
