"""STRING_DIFF / HEX_PACKING encodings.

Restatements of ObStringDiffEncoder (ob_string_diff_encoder.{h,cpp}) and
ObHexStringEncoder (ob_hex_string_encoder.{h,cpp}) for fixed char(N<=8)
columns: byte-level roundtrip via the oracle AND the independent Python
format model, filter/group parity oracle-vs-GPU through the generic
(slow-decode) kernel path.
"""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from oceanbase_amd import abi, oracle  # noqa: E402
import pymodel  # noqa: E402


def _char_col(rows, seed, common="AB", varying=(b"abcdefgh", b"0123"),
              nulls_at=()):
    """4-char strings: positions 0,2 fixed, 1,3 drawn from small alphabets."""
    rng = np.random.default_rng(seed)
    a = np.empty((rows, 4), dtype=np.uint8)
    a[:, 0] = ord(common[0])
    a[:, 2] = ord(common[1])
    a[:, 1] = rng.choice(np.frombuffer(varying[0], dtype=np.uint8), rows)
    a[:, 3] = rng.choice(np.frombuffer(varying[1], dtype=np.uint8), rows)
    nb = np.zeros((rows + 7) // 8, dtype=np.uint8)
    for r in nulls_at:
        nb[r // 8] |= 1 << (r % 8)
    return a, nb


def _schema4():
    return oracle.make_schema([(abi.T_CHAR, 0, 0, 4)])


@pytest.mark.parametrize("enc", [abi.ENC_SDIFF, abi.ENC_HEX])
@pytest.mark.parametrize("with_nulls", [False, True])
def test_roundtrip_oracle_and_pymodel(enc, with_nulls):
    rows = 903
    nulls_at = (0, 17, 900) if with_nulls else ()
    a, nb = _char_col(rows, seed=5, nulls_at=nulls_at)
    schema = _schema4()
    blk = oracle.encode_block(schema, [a.reshape(-1)], [enc],
                              [nb if with_nulls else None])
    # oracle decode
    rc, outs, nbs = oracle.decode_block(schema, 1, blk, [0])
    got = np.frombuffer(outs[0], dtype=np.uint8).reshape(rows, 4)
    for r in range(rows):
        isn = (nbs[0][r // 8] >> (r % 8)) & 1
        if r in nulls_at:
            assert isn
        else:
            assert not isn and (got[r] == a[r]).all()
    # independent python model decode (byte-format pin)
    pb = pymodel.Block(blk, [(abi.T_CHAR, 0, 0, 4)])
    vals = pb.decode_col(0)
    for r in range(rows):
        if r in nulls_at:
            assert vals[r] is None
        else:
            assert vals[r] == int.from_bytes(a[r].tobytes(), "little")


def test_string_diff_hex_packs_diff_bytes():
    """When the varying bytes use <=16 distinct chars, STRING_DIFF must
    nibble-pack them (is_hex_packing, ob_string_diff_encoder.h:80-83):
    the block must be smaller than the plain-diff equivalent."""
    rows = 1000
    a, _ = _char_col(rows, seed=9)
    schema = _schema4()
    blk = oracle.encode_block(schema, [a.reshape(-1)], [abi.ENC_SDIFF], None)
    # 2 diff bytes/row hex-packed -> 1 B/row (+meta) vs 4 B/row RAW
    raw = oracle.encode_block(schema, [a.reshape(-1)], [abi.ENC_RAW], None)
    assert len(blk) < len(raw) - rows  # at least 1 B/row saved over RAW


@pytest.mark.parametrize("enc", [abi.ENC_SDIFF, abi.ENC_HEX])
def test_filter_parity_cpu(enc):
    rows = 1200
    a, nb = _char_col(rows, seed=13, nulls_at=(3, 700))
    schema = _schema4()
    blk = oracle.encode_block(schema, [a.reshape(-1)], [enc], [nb])
    # char EQ / range filters (byte-lexicographic)
    for op, lo in ((abi.OP_EQ, int.from_bytes(b"Aa" + b"B0", "little")),
                   (abi.OP_GE, int.from_bytes(b"AdB2", "little")),
                   (abi.OP_NN, 0), (abi.OP_NU, 0)):
        filt = abi.make_filter([dict(col=0, op=op, lo=lo)])
        bits, passed = oracle.filter_block(schema, 1, blk, filt)
        # independent count via pymodel
        pb = pymodel.Block(blk, [(abi.T_CHAR, 0, 0, 4)])
        vals = pb.decode_col(0)
        want = 0
        for v in vals:
            want += pymodel.eval_leaf(op, v, lo, 0, [], pymodel.SC_STRING, 4)
        assert passed == want


@pytest.mark.gpu
@pytest.mark.parametrize("enc", [abi.ENC_SDIFF, abi.ENC_HEX])
def test_gpu_parity_filter_and_group(enc):
    """GPU generic path (slow decode) vs oracle: filter + group-by over a
    string-transform encoded char column."""
    from oceanbase_amd.engine import GpuEngine
    from test_gpu_parity import _manual_blockset
    rows = 3000
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, 4), (abi.T_INT, 0, 0, 8)])
    a, nb = _char_col(rows, seed=21, varying=(b"abcd", b"01"),
                      nulls_at=(10, 2000))  # 8 combos + null: fits the
    # per-workgroup LDS group table (16 slots) on the generic path
    rng = np.random.default_rng(22)
    q = rng.integers(1, 1000, rows, dtype=np.int64)
    blocks = []
    for s in range(0, rows, 1000):
        e = min(s + 1000, rows)
        nb_w = np.zeros((e - s + 7) // 8, dtype=np.uint8)
        for r in (10, 2000):
            if s <= r < e:
                nb_w[(r - s) // 8] |= 1 << ((r - s) % 8)
        blocks.append(oracle.encode_block(
            schema, [a[s:e].reshape(-1), q[s:e].view(np.uint8)],
            [enc, abi.ENC_RAW], [nb_w, None]))
    bs = _manual_blockset(schema, blocks)
    bs.total_rows = rows
    eng = GpuEngine(0)
    h = eng.load(bs)
    filt = abi.make_filter(
        [dict(col=0, op=abi.OP_GE, lo=int.from_bytes(b"AcB0", "little"))])
    agg = abi.make_agg([0], [dict(kind=abi.AGG_COUNT),
                             dict(kind=abi.AGG_SUM, col_a=1)])
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    assert sorted(abi.result_rows(res_gpu, 2)) == sorted(
        abi.result_rows(res_cpu, 2))
    # pure filter too (bitmap/row-id path)
    assert eng.filter(h, filt) == res_cpu.rows_passed
    eng.close()


@pytest.mark.gpu
def test_jit_group_spill_past_wg_table():
    """>16 distinct groups through the JIT path: per-block dict products
    stay small (eligible) but the union exceeds the 16-slot workgroup
    table, exercising the cold global-table spill pass."""
    from oceanbase_amd.engine import GpuEngine
    from test_gpu_parity import _manual_blockset
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, 2), (abi.T_INT, 0, 0, 8)])
    rng = np.random.default_rng(31)
    blocks = []
    all_keys, all_q = [], []
    for bi in range(8):  # 8 blocks x 4 disjoint keys = 32 groups
        rows = 600
        keys = [bytes([65 + bi, 97 + j]) for j in range(4)]
        pick = rng.integers(0, 4, rows)
        a = np.frombuffer(b"".join(keys[p] for p in pick), dtype=np.uint8)
        q = rng.integers(1, 100, rows, dtype=np.int64)
        all_keys.extend(keys[p] for p in pick)
        all_q.append(q)
        blocks.append(oracle.encode_block(
            schema, [a.copy(), q.view(np.uint8)],
            [abi.ENC_DICT, abi.ENC_RAW], None))
    bs = _manual_blockset(schema, blocks)
    bs.total_rows = 8 * 600
    eng = GpuEngine(0)
    h = eng.load(bs)
    agg = abi.make_agg([0], [dict(kind=abi.AGG_COUNT),
                             dict(kind=abi.AGG_SUM, col_a=1)])
    res_gpu = eng.scan_filter_agg(h, None, agg)
    assert eng.last_jit(), "expected the JIT path for this plan shape"
    res_cpu = oracle.scan_filter_agg(bs, None, agg)
    assert res_gpu.n_groups == 32 == res_cpu.n_groups
    assert sorted(abi.result_rows(res_gpu, 2)) == sorted(
        abi.result_rows(res_cpu, 2))
    eng.close()


def test_hex_nibble_order_matches_reference():
    """Recreates TEST(ObHexStringMap, store_order) from the reference's
    unittest/storage/blocksstable/encoding/test_hex.cpp:17-40: packing the
    digits '0'..'9' and hex-printing the packed bytes must reproduce the
    input — pinning both the ascending-char index map and the
    high-nibble-first pack order."""
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, 8)])
    a = np.frombuffer(b"01234567", dtype=np.uint8).copy()
    blk = oracle.encode_block(schema, [a], [abi.ENC_HEX], None)
    pb = pymodel.Block(blk, [(abi.T_CHAR, 0, 0, 8)])
    ch = pb.col_headers[0]
    base = pb.meta_base + ch["offset"]
    import struct
    ver, nch, ssize = struct.unpack_from("<BBH", pb.data, base)
    assert nch == 8 and ssize == 8
    chars = bytes(pb.data[base + 4:base + 4 + nch])
    assert chars == b"01234567"  # build_index: ascending char order
    row = bytes(pb.data[base + ch["length"]:base + ch["length"] + 4])
    # '0'..'7' map to nibbles 0..7; high nibble first => hex print equals
    # the input string, exactly as the reference test asserts
    assert row.hex() == "01234567"


# ---- COLUMN_EQUAL (span encoder: this column == a previous column except
# at exception rows; ObColumnEqualEncoder, ob_column_equal_encoder.h) ----

def _coleq_blockset(rows=3000, seed=41, nulls=True):
    schema = oracle.make_schema([(abi.T_INT, 0, 0, 8), (abi.T_INT, 0, 0, 8)])
    rng = np.random.default_rng(seed)
    a = rng.integers(-10**9, 10**9, rows)
    b = a.copy()
    exc = rng.choice(rows, rows // 40, replace=False)
    b[exc] += rng.integers(1, 100, len(exc))
    na = np.zeros((rows + 7) // 8, dtype=np.uint8)
    nb = np.zeros((rows + 7) // 8, dtype=np.uint8)
    if nulls:
        na[0] |= 1      # row 0: A null, B not -> exception
        nb[1] |= 1      # row 8: B null only -> null exception
    return schema, a, b, na, nb


def test_column_equal_roundtrip_and_pymodel():
    schema, a, b, na, nb = _coleq_blockset()
    rows = len(a)
    blk = oracle.encode_block(schema, [a.view(np.uint8), b.view(np.uint8)],
                              [abi.ENC_RAW, abi.ENC_COLUMN_EQUAL], [na, nb])
    rc, outs, nbs = oracle.decode_block(schema, 2, blk, [1])
    got = np.frombuffer(outs[0], dtype=np.int64)
    for r in range(rows):
        isn = (nbs[0][r // 8] >> (r % 8)) & 1
        want_null = bool((nb[r // 8] >> (r % 8)) & 1)
        assert bool(isn) == want_null
        if not isn:
            assert got[r] == b[r]
    pb = pymodel.Block(blk, [(abi.T_INT, 0, 0, 8), (abi.T_INT, 0, 0, 8)])
    vals = pb.decode_col(1)
    for r in range(rows):
        want = None if (nb[r // 8] >> (r % 8)) & 1 else int(b[r])
        assert vals[r] == want


def test_column_equal_filter_and_agg_cpu():
    schema, a, b, na, nb = _coleq_blockset()
    blk = oracle.encode_block(schema, [a.view(np.uint8), b.view(np.uint8)],
                              [abi.ENC_RAW, abi.ENC_COLUMN_EQUAL], [na, nb])
    filt = abi.make_filter([dict(col=1, op=abi.OP_GT, lo=0)])
    bits, passed = oracle.filter_block(schema, 2, blk, filt)
    want = sum(1 for r in range(len(b))
               if not (nb[r // 8] >> (r % 8)) & 1 and b[r] > 0)
    assert passed == want


@pytest.mark.gpu
def test_column_equal_gpu_parity():
    from oceanbase_amd.engine import GpuEngine
    from test_gpu_parity import _manual_blockset
    schema, a, b, na, nb = _coleq_blockset()
    rows = len(a)
    blocks = []
    for s in range(0, rows, 1000):
        e = min(s + 1000, rows)
        naw = np.zeros((e - s + 7) // 8, dtype=np.uint8)
        nbw = np.zeros((e - s + 7) // 8, dtype=np.uint8)
        for r in range(s, e):
            if (na[r // 8] >> (r % 8)) & 1:
                naw[(r - s) // 8] |= 1 << ((r - s) % 8)
            if (nb[r // 8] >> (r % 8)) & 1:
                nbw[(r - s) // 8] |= 1 << ((r - s) % 8)
        blocks.append(oracle.encode_block(
            schema, [a[s:e].view(np.uint8), b[s:e].view(np.uint8)],
            [abi.ENC_RAW, abi.ENC_COLUMN_EQUAL], [naw, nbw]))
    bs = _manual_blockset(schema, blocks)
    bs.total_rows = rows
    eng = GpuEngine(0)
    h = eng.load(bs)
    # filter on the COLUMN_EQUAL column + SUM over it
    filt = abi.make_filter([dict(col=1, op=abi.OP_GT, lo=0)])
    agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT),
                            dict(kind=abi.AGG_SUM, col_a=1)])
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    assert sorted(abi.result_rows(res_gpu, 2)) == sorted(
        abi.result_rows(res_cpu, 2))
    assert eng.filter(h, filt) == res_cpu.rows_passed
    eng.close()


# ---- STRING_PREFIX (prefix table + per-row ref/suffix;
# ObStringPrefixEncoder, ob_string_prefix_encoder.h) ----

def _prefix_col(rows, seed, prefixes=(b"ABC", b"XYZ", b"MNO"),
                alpha=b"0123456789", nulls_at=()):
    rng = np.random.default_rng(seed)
    n = len(prefixes[0]) + 3
    a = np.zeros((rows, n), dtype=np.uint8)
    for r in range(rows):
        p = prefixes[rng.integers(0, len(prefixes))]
        sfx = bytes(rng.choice(np.frombuffer(alpha, dtype=np.uint8), 3))
        a[r] = np.frombuffer(p + sfx, dtype=np.uint8)
    nb = np.zeros((rows + 7) // 8, dtype=np.uint8)
    for r in nulls_at:
        nb[r // 8] |= 1 << (r % 8)
    return a, nb, n


@pytest.mark.parametrize("with_nulls", [False, True])
def test_string_prefix_roundtrip_and_pymodel(with_nulls):
    rows = 900
    nulls_at = (1, 500) if with_nulls else ()
    a, nb, n = _prefix_col(rows, seed=3, nulls_at=nulls_at)
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, n)])
    blk = oracle.encode_block(schema, [a.reshape(-1)],
                              [abi.ENC_STRING_PREFIX],
                              [nb if with_nulls else None])
    rc, outs, nbs = oracle.decode_block(schema, 1, blk, [0])
    got = np.frombuffer(outs[0], dtype=np.uint8).reshape(rows, n)
    for r in range(rows):
        isn = (nbs[0][r // 8] >> (r % 8)) & 1
        if r in nulls_at:
            assert isn
        else:
            assert not isn and (got[r] == a[r]).all()
    pb = pymodel.Block(blk, [(abi.T_CHAR, 0, 0, n)])
    vals = pb.decode_col(0)
    for r in range(rows):
        want = (None if r in nulls_at
                else int.from_bytes(a[r].tobytes(), "little"))
        assert vals[r] == want


@pytest.mark.gpu
def test_string_prefix_gpu_parity():
    from oceanbase_amd.engine import GpuEngine
    from test_gpu_parity import _manual_blockset
    rows = 2400
    a, nb, n = _prefix_col(rows, seed=23, nulls_at=(7, 1800))
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, n),
                                 (abi.T_INT, 0, 0, 8)])
    rng = np.random.default_rng(24)
    q = rng.integers(1, 500, rows, dtype=np.int64)
    blocks = []
    for s in range(0, rows, 800):
        e = s + 800
        nbw = np.zeros((e - s + 7) // 8, dtype=np.uint8)
        for r in (7, 1800):
            if s <= r < e:
                nbw[(r - s) // 8] |= 1 << ((r - s) % 8)
        blocks.append(oracle.encode_block(
            schema, [a[s:e].reshape(-1), q[s:e].view(np.uint8)],
            [abi.ENC_STRING_PREFIX, abi.ENC_RAW], [nbw, None]))
    bs = _manual_blockset(schema, blocks)
    bs.total_rows = rows
    eng = GpuEngine(0)
    h = eng.load(bs)
    lo = int.from_bytes(b"MNO000", "little")
    filt = abi.make_filter([dict(col=0, op=abi.OP_GE, lo=lo)])
    agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT),
                            dict(kind=abi.AGG_SUM, col_a=1)])
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    assert sorted(abi.result_rows(res_gpu, 2)) == sorted(
        abi.result_rows(res_cpu, 2))
    assert eng.filter(h, filt) == res_cpu.rows_passed
    eng.close()


# ---- COLUMN_SUBSTR (this column is a fixed slice of a previous char
# column except at exception rows; ObInterColSubStrEncoder) ----

def _substr_blockset(rows=2100, seed=51):
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, 6), (abi.T_CHAR, 0, 0, 3)])
    rng = np.random.default_rng(seed)
    alpha = np.frombuffer(b"ABCDEFGH", dtype=np.uint8)
    a = rng.choice(alpha, (rows, 6))
    b = a[:, 2:5].copy()
    for r in rng.choice(rows, rows // 50, replace=False):
        b[r, 0] ^= 1
    na = np.zeros((rows + 7) // 8, dtype=np.uint8); na[0] |= 1
    nb = np.zeros((rows + 7) // 8, dtype=np.uint8); nb[0] |= 3
    return schema, a, b, na, nb


def test_column_substr_roundtrip_and_pymodel():
    schema, a, b, na, nb = _substr_blockset()
    rows = len(a)
    blk = oracle.encode_block(schema, [a.reshape(-1), b.reshape(-1)],
                              [abi.ENC_RAW, abi.ENC_COLUMN_SUBSTR], [na, nb])
    rc, outs, nbs = oracle.decode_block(schema, 2, blk, [1])
    got = np.frombuffer(outs[0], dtype=np.uint8).reshape(rows, 3)
    for r in range(rows):
        isn = (nbs[0][r // 8] >> (r % 8)) & 1
        assert bool(isn) == bool((nb[r // 8] >> (r % 8)) & 1)
        if not isn:
            assert (got[r] == b[r]).all()
    pb = pymodel.Block(blk, [(abi.T_CHAR, 0, 0, 6), (abi.T_CHAR, 0, 0, 3)])
    vals = pb.decode_col(1)
    for r in range(rows):
        want = (None if (nb[r // 8] >> (r % 8)) & 1
                else int.from_bytes(b[r].tobytes(), "little"))
        assert vals[r] == want


@pytest.mark.gpu
def test_column_substr_gpu_parity():
    from oceanbase_amd.engine import GpuEngine
    from test_gpu_parity import _manual_blockset
    schema, a, b, na, nb = _substr_blockset()
    rows = len(a)
    blocks = []
    for s in range(0, rows, 700):
        e = min(s + 700, rows)
        naw = np.zeros((e - s + 7) // 8, dtype=np.uint8)
        nbw = np.zeros((e - s + 7) // 8, dtype=np.uint8)
        for r in range(s, e):
            if (na[r // 8] >> (r % 8)) & 1:
                naw[(r - s) // 8] |= 1 << ((r - s) % 8)
            if (nb[r // 8] >> (r % 8)) & 1:
                nbw[(r - s) // 8] |= 1 << ((r - s) % 8)
        blocks.append(oracle.encode_block(
            schema, [a[s:e].reshape(-1), b[s:e].reshape(-1)],
            [abi.ENC_RAW, abi.ENC_COLUMN_SUBSTR], [naw, nbw]))
    bs = _manual_blockset(schema, blocks)
    bs.total_rows = rows
    eng = GpuEngine(0)
    h = eng.load(bs)
    lo = int.from_bytes(b"D" + b"\x00\x00", "little")
    filt = abi.make_filter([dict(col=1, op=abi.OP_GE, lo=lo)])
    agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT)])
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    assert eng.filter(h, filt) == res_cpu.rows_passed
    eng.close()
