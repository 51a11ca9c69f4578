"""GPU-native CS load parity (product half).

The product path decodes CS micro blocks ON DEVICE at load time
(obx_gpu_load_cs_blocks: host metadata parse + obx_cs_kernels.hip stream
decode — the GPU-native ObCSMicroBlockTransformer). These tests compare
the loaded handle's decoded columns and query results bit-exactly against
the CPU oracle's CS decode (oracle/obx_cs_block.c via test_cs_block's
helpers). liboracle.so never runs in the product path — it is the checker.
"""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oceanbase_amd import abi, oracle  # noqa: E402
from test_cs_block import (  # noqa: E402
    _enc, _dec, _get_int, _get_str, _int_col, _str_col)


@pytest.fixture
def eng():
    from oceanbase_amd.engine import GpuEngine
    e = GpuEngine(0)
    yield e
    e.close()


def _load(eng, blocks, specs):
    schema = oracle.make_schema(specs)
    return eng.load_cs(blocks, schema), schema


def _fetch_int(eng, h, col, rows):
    eng.decode(h, [col])
    return eng.fetch_col(h, col, 8).view(np.int64)


def _fetch_bytes(eng, h, col, rows, length):
    eng.decode(h, [col])
    return eng.fetch_col(h, col, length)


def _nn_count(eng, h, col):
    filt = abi.make_filter([dict(col=col, op=abi.OP_NN)])
    return eng.filter(h, filt)


CODECS = [1, 2, 3, 4, 5, 6, 8]  # RAW, DDZR, DDZP, DZR, DZP, FPFOR, XPFOR


@pytest.mark.gpu
@pytest.mark.parametrize("enc", CODECS)
def test_int_codecs_match_oracle(eng, enc):
    rng = np.random.default_rng(enc * 100 + 7)
    rows = 1500
    # shapes that exercise deltas, runs and exceptions
    base = np.cumsum(rng.integers(-40, 45, rows)).astype(np.int64)
    base[::97] += 1 << 33  # outliers -> PFoR exceptions
    blob = _enc(rows, [_int_col(base, enc=enc)])
    v = _dec(blob)
    exp, exp_nulls = _get_int(v, 0)
    assert not exp_nulls
    h, _ = _load(eng, [blob], [(abi.T_INT, 0, 19, 8)])
    got = _fetch_int(eng, h, 0, rows)
    assert np.array_equal(got, exp), f"codec {enc} mismatch"
    eng.free(h)


@pytest.mark.gpu
@pytest.mark.parametrize("enc", [1, 5, 6])
def test_int_nulls_replace_and_bitmap(eng, enc):
    rng = np.random.default_rng(31 + enc)
    rows = 1200
    vals = rng.integers(-1000, 1000, rows).astype(np.int64)
    null_rows = sorted(rng.choice(rows, 60, replace=False).tolist())
    blob = _enc(rows, [_int_col(vals, null_rows=null_rows, enc=enc)])
    v = _dec(blob)
    exp, exp_nulls = _get_int(v, 0)
    assert exp_nulls == set(null_rows)
    h, _ = _load(eng, [blob], [(abi.T_INT, 0, 19, 8)])
    got = _fetch_int(eng, h, 0, rows)
    assert np.array_equal(got, exp)
    assert _nn_count(eng, h, 0) == rows - len(null_rows)
    eng.free(h)

    # bitmap-null form: values spanning the full int64 range force the
    # column layer to a bitmap (no adjacent replace value exists)
    vals2 = vals.copy()
    vals2[0] = np.iinfo(np.int64).min
    vals2[1] = np.iinfo(np.int64).max
    blob2 = _enc(rows, [_int_col(vals2, null_rows=null_rows, enc=1)])
    v2 = _dec(blob2)
    exp2, exp2_nulls = _get_int(v2, 0)
    assert exp2_nulls == set(null_rows)
    h2, _ = _load(eng, [blob2], [(abi.T_INT, 0, 19, 8)])
    got2 = _fetch_int(eng, h2, 0, rows)
    assert np.array_equal(got2, exp2)
    assert _nn_count(eng, h2, 0) == rows - len(null_rows)
    eng.free(h2)


@pytest.mark.gpu
def test_int_dict_and_const_ref(eng):
    rng = np.random.default_rng(5)
    rows = 2000
    # plain dict with nulls
    vals = rng.choice(np.array([3, 11, 42, 77, 500, -9], dtype=np.int64),
                      rows)
    null_rows = sorted(rng.choice(rows, 40, replace=False).tolist())
    blob = _enc(rows, [_int_col(vals, null_rows=null_rows, dict_=True)])
    v = _dec(blob)
    exp, exp_nulls = _get_int(v, 0)
    h, _ = _load(eng, [blob], [(abi.T_INT, 0, 19, 8)])
    got = _fetch_int(eng, h, 0, rows)
    assert np.array_equal(got, exp)
    assert _nn_count(eng, h, 0) == rows - len(exp_nulls)
    eng.free(h)

    # dominant value -> CONST_ENCODING_REF exception layout
    vals2 = np.full(rows, 1234, dtype=np.int64)
    exc = rng.choice(rows, 30, replace=False)
    vals2[exc] = rng.integers(1, 1000, 30)
    blob2 = _enc(rows, [_int_col(vals2, dict_=True)])
    v2 = _dec(blob2)
    exp2, _n2 = _get_int(v2, 0)
    h2, _ = _load(eng, [blob2], [(abi.T_INT, 0, 19, 8)])
    got2 = _fetch_int(eng, h2, 0, rows)
    assert np.array_equal(got2, exp2)
    eng.free(h2)


@pytest.mark.gpu
def test_fixed_string_and_str_dict(eng):
    rng = np.random.default_rng(9)
    rows = 1100
    alpha = [b"AAAA", b"BBBB", b"CCCC", b"DDDD"]
    strings = [alpha[int(i)] for i in rng.integers(0, 4, rows)]
    null_rows = sorted(rng.choice(rows, 25, replace=False).tolist())
    blob = _enc(rows, [_str_col(strings, null_rows=null_rows)])
    v = _dec(blob)
    exp, exp_nulls = _get_str(v, 0)
    h, _ = _load(eng, [blob], [(abi.T_CHAR, 0, 0, 4)])
    got = _fetch_bytes(eng, h, 0, rows, 4)
    for r in range(rows):
        cell = bytes(got[r * 4:(r + 1) * 4])
        if r in exp_nulls:
            continue  # engine zero-fills null cells
        assert cell == exp[r], r
    assert _nn_count(eng, h, 0) == rows - len(null_rows)
    eng.free(h)

    blob2 = _enc(rows, [_str_col(strings, null_rows=null_rows, dict_=True)])
    v2 = _dec(blob2)
    exp2, exp2_nulls = _get_str(v2, 0)
    h2, _ = _load(eng, [blob2], [(abi.T_CHAR, 0, 0, 4)])
    got2 = _fetch_bytes(eng, h2, 0, rows, 4)
    for r in range(rows):
        if r in exp2_nulls:
            continue
        assert bytes(got2[r * 4:(r + 1) * 4]) == exp2[r], r
    assert _nn_count(eng, h2, 0) == rows - len(exp2_nulls)
    eng.free(h2)


@pytest.mark.gpu
def test_multi_chunk_block(eng):
    """A CS block larger than one LDS-stageable chunk splits into several
    dev_blocks; values and row order must survive the split."""
    rng = np.random.default_rng(3)
    rows = 9000  # int64 col -> 72 KB decoded -> several chunks
    vals = rng.integers(-10**9, 10**9, rows).astype(np.int64)
    blob = _enc(rows, [_int_col(vals, enc=6)])
    v = _dec(blob)
    exp, _ = _get_int(v, 0)
    h, _ = _load(eng, [blob], [(abi.T_INT, 0, 19, 8)])
    got = _fetch_int(eng, h, 0, rows)
    assert np.array_equal(got, exp)
    eng.free(h)


@pytest.mark.gpu
def test_cs_q1_style_end_to_end(eng):
    """Q1-shaped query over a CS-encoded table: group-by char dict +
    decimal-style product sums, GPU CS handle vs the oracle pipeline over
    the transcoded PAX form (cross-format equivalence)."""
    import cs_oracle_util as csu
    rng = np.random.default_rng(17)
    rows_per_block, nblocks = 3000, 4
    blocks = []
    all_cols = {k: [] for k in range(5)}
    for b in range(nblocks):
        qty = rng.integers(1, 51, rows_per_block).astype(np.int64)
        price = rng.integers(90000, 10**7, rows_per_block).astype(np.int64)
        disc = rng.integers(0, 11, rows_per_block).astype(np.int64)
        flag = [b"A" if x == 0 else b"N" if x == 1 else b"R"
                for x in rng.integers(0, 3, rows_per_block)]
        date = rng.integers(8000, 10600, rows_per_block).astype(np.int64)
        blocks.append(_enc(rows_per_block, [
            _int_col(qty, dict_=True),
            _int_col(price, enc=6),
            _int_col(disc, dict_=True),
            _str_col(flag, dict_=True),
            _int_col(date, enc=5),
        ]))
        for k, vco in enumerate((qty, price, disc, flag, date)):
            all_cols[k].append(vco)
    # scales 0 on both sides: the oracle leg re-encodes through
    # to_pax_blocks, whose schema is scale-0 T_INT — one_b must agree
    specs = [(abi.T_INT, 0, 19, 8), (abi.T_INT, 0, 19, 8),
             (abi.T_INT, 0, 19, 8), (abi.T_CHAR, 0, 0, 1),
             (abi.T_INT, 0, 19, 8)]
    h, _schema = _load(eng, blocks, specs)
    filt = abi.make_filter([dict(col=4, op=abi.OP_LE, lo=10471)])
    agg = abi.make_agg([3], [
        dict(kind=abi.AGG_COUNT),
        dict(kind=abi.AGG_SUM, col_a=0),
        dict(kind=abi.AGG_SUM, col_a=1),
        dict(kind=abi.AGG_SUM_PROD2, col_a=1, col_b=2),
    ])
    res_gpu = eng.scan_filter_agg(h, filt, agg)

    # oracle side: same table through the CS->PAX transcode + CPU pipeline
    schema2, pax = csu.to_pax_blocks(
        blocks, declared_specs=[(abi.T_INT, 0, 19, 8)] * 3 +
        [(abi.T_CHAR, 0, 0, 1), (abi.T_INT, 0, 19, 8)])
    from test_cs_pipeline_equiv import _pax_blockset
    bs = _pax_blockset(schema2, pax)
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    assert sorted(abi.result_rows(res_gpu, 4)) == sorted(
        abi.result_rows(res_cpu, 4))
    eng.free(h)


def test_cs_host_parse_cpu():
    """The product CS header probe works without a GPU (loadable ABI)."""
    import ctypes as C
    rows = 64
    blob = _enc(rows, [_int_col(np.arange(rows, dtype=np.int64))])
    lib = C.CDLL(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "oceanbase_amd", "libobx.so"))
    lib.obx_cs_host_parse.restype = C.c_int
    lib.obx_cs_host_parse.argtypes = [C.c_char_p, C.c_int64,
                                      C.POINTER(C.c_uint32),
                                      C.POINTER(C.c_uint32)]
    r = C.c_uint32()
    n = C.c_uint32()
    assert lib.obx_cs_host_parse(blob, len(blob), C.byref(r),
                                 C.byref(n)) == 0
    assert r.value == rows and n.value == 1
