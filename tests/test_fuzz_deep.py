"""Deeper seeded fuzzing (CPU-only surfaces).

1. Black-filter programs: 60 random valid postfix programs over random
   multi-column data — oracle bitmap vs the independent python evaluator.
2. CS integer codecs vs the reference binaries (when oracle/_ref is
   built): 150 random (codec, width, count, distribution) shapes,
   byte-identical encode + reference-decodes-our-bytes.
"""
import ctypes as C
import os
import random

import numpy as np
import pytest

from oceanbase_amd import abi, oracle
from test_black_filter import py_black_eval

M64 = (1 << 64) - 1
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _rand_prog(rng, n_cols, n_consts):
    """Random postfix program ending with a comparison/logic root."""
    arith = [abi.BX_ADD, abi.BX_SUB, abi.BX_MUL, abi.BX_DIV,
             abi.BX_MOD]
    cmps = [abi.BX_LT, abi.BX_LE, abi.BX_GT, abi.BX_GE, abi.BX_EQ,
            abi.BX_NE]
    prog = []
    depth = 0
    # build 1-3 comparison terms, then AND/OR them
    n_terms = rng.randint(1, 3)
    for _ in range(n_terms):
        # each side: col/const with optional arith
        for _side in range(2):
            prog.append(rng.choice(
                [abi.BX_COL | rng.randrange(n_cols),
                 abi.BX_CONST | rng.randrange(n_consts)]))
            if rng.random() < 0.4:
                prog.append(rng.choice(
                    [abi.BX_COL | rng.randrange(n_cols),
                     abi.BX_CONST | rng.randrange(n_consts)]))
                prog.append(rng.choice(arith))
            if rng.random() < 0.15:
                prog.append(abi.BX_NEG)
        prog.append(rng.choice(cmps))
        depth += 1
    while depth > 1:
        prog.append(rng.choice([abi.BX_AND, abi.BX_OR]))
        depth -= 1
    if rng.random() < 0.2:
        prog.append(abi.BX_NOT)
    return prog


def test_black_program_fuzz_oracle_vs_pymodel():
    rng = random.Random(424242)
    nrng = np.random.default_rng(11)
    rows = 800
    n_cols, n_consts = 3, 3
    data = [nrng.integers(-1000, 1000, rows).astype(np.int64)
            for _ in range(n_cols)]
    null_rows = set(range(0, rows, 17))
    nulls = np.zeros((rows + 7) // 8, dtype=np.uint8)
    for r in null_rows:
        nulls[r >> 3] |= 1 << (r & 7)
    schema = oracle.make_schema([(abi.T_INT, 0, 0, 8)] * n_cols)
    blk = oracle.encode_block(
        schema, [d.view(np.uint8) for d in data],
        [abi.ENC_RAW] * n_cols, [nulls, None, None])
    n_checked = 0
    for i in range(60):
        prog = _rand_prog(rng, n_cols, n_consts)
        if len(prog) > abi.__dict__.get("BX_MAX_PROG", 24) and len(prog) > 24:
            continue
        consts = [rng.randint(-500, 500) for _ in range(n_consts)]
        filt = abi.make_filter([dict(op=abi.OP_BLACK,
                                     bcols=list(range(n_cols)),
                                     bconst=consts, bprog=prog)])
        bits, pc = oracle.filter_block(schema, n_cols, blk, filt)
        exp = 0
        for r in range(rows):
            vals = [int(d[r]) for d in data]
            nl = [r in null_rows, False, False]
            want = py_black_eval(prog, consts, vals, nl)
            got = bool(bits[r >> 3] & (1 << (r & 7)))
            assert got == want, (i, r, prog)
            exp += want
        assert pc == exp
        n_checked += 1
    assert n_checked >= 50


_ref_so = os.path.join(REPO, "oracle", "_ref", "librefcodec.so")
needs_ref = pytest.mark.skipif(not os.path.exists(_ref_so),
                               reason="oracle/_ref not built here")


@needs_ref
def test_cs_codec_fuzz_vs_reference():
    from test_ref_parity import _codec_libs, CODECS
    ref, ours = _codec_libs()
    rng = np.random.default_rng(20260915)
    FN = dict((e, f) for f, e in CODECS)
    n = 0
    for _ in range(150):
        enc_type = int(rng.choice(list(FN.keys())))
        wb = int(rng.choice([1, 2, 4, 8]))
        cnt = int(rng.integers(1, 700))
        if enc_type == 6:
            # the reference SIMD codec pads to whole 128-value frames;
            # stream encoders only select it for full frames
            cnt = max(128, (cnt // 128) * 128)
        mask = (1 << (8 * wb)) - 1
        kind = rng.integers(0, 4)
        if kind == 0:      # random
            vals = rng.integers(0, 1 << min(8 * wb, 63), cnt,
                                dtype=np.uint64) & mask
        elif kind == 1:    # ramp with jitter
            vals = (np.arange(cnt, dtype=np.uint64) * 3 +
                    rng.integers(0, 5, cnt, dtype=np.uint64)) & mask
        elif kind == 2:    # runs
            vals = np.repeat(rng.integers(0, 100, max(cnt // 7, 1),
                                          dtype=np.uint64),
                             7)[:cnt].copy() & mask
            if len(vals) < cnt:
                vals = np.pad(vals, (0, cnt - len(vals)))
        else:              # small range
            vals = rng.integers(0, 17, cnt, dtype=np.uint64) & mask
        raw = vals.astype(f"<u{wb}").tobytes()
        f = getattr(ours, FN[enc_type])
        f.restype = C.c_int64
        f.argtypes = [C.c_char_p, C.c_uint32, C.c_uint32, C.c_char_p,
                      C.c_size_t]
        cap = len(raw) * 3 + 8192
        b1 = C.create_string_buffer(cap)
        b2 = C.create_string_buffer(cap)
        n1 = f(raw, cnt, wb, b1, cap)
        n2 = ref.ref_codec_encode(enc_type, raw, cnt, wb, b2, cap)
        assert n1 == n2 and b1.raw[:n1] == b2.raw[:n2], (enc_type, wb, cnt,
                                                         int(kind))
        if not (enc_type == 6 and cnt % 128):
            dec = C.create_string_buffer(len(raw))
            m = ref.ref_codec_decode(enc_type, b1.raw[:n1], n1, cnt, wb,
                                     dec, len(raw))
            assert m == n1 and dec.raw == raw, (enc_type, wb, cnt)
        n += 1
    assert n == 150


@pytest.mark.gpu
def test_black_program_fuzz_gpu_vs_oracle():
    """The device black VM across 30 random programs (multi-block set,
    NULLs included): survivor counts + bitmaps vs the oracle."""
    from oceanbase_amd.engine import GpuEngine
    rng = random.Random(777)
    nrng = np.random.default_rng(23)
    rows_pb, nblocks = 2000, 5
    n_cols, n_consts = 3, 3
    schema = oracle.make_schema([(abi.T_INT, 0, 0, 8)] * n_cols)
    blocks = []
    for _ in range(nblocks):
        data = [nrng.integers(-1000, 1000, rows_pb).astype(np.int64)
                for _ in range(n_cols)]
        nulls = np.zeros((rows_pb + 7) // 8, dtype=np.uint8)
        for r in range(0, rows_pb, 13):
            nulls[r >> 3] |= 1 << (r & 7)
        blocks.append(oracle.encode_block(
            schema, [d.view(np.uint8) for d in data],
            [abi.ENC_RAW] * n_cols, [nulls, None, None]))
    import test_group_capacity as tgc
    bs = tgc._blockset(schema, blocks, rows_pb * nblocks)
    eng = GpuEngine()
    h = eng.load(bs)
    checked = 0
    for i in range(30):
        prog = _rand_prog(rng, n_cols, n_consts)
        if len(prog) > 24:
            continue
        consts = [rng.randint(-500, 500) for _ in range(n_consts)]
        filt = abi.make_filter([dict(op=abi.OP_BLACK,
                                     bcols=list(range(n_cols)),
                                     bconst=consts, bprog=prog)])
        res_cpu = oracle.scan_filter_agg(bs, filt, None)
        survivors = eng.filter(h, filt)
        assert survivors == res_cpu.rows_passed, (i, prog)
        checked += 1
    assert checked >= 25
    eng.free(h)


def test_pax_corruption_fuzz_never_crashes():
    """200 random single-byte corruptions of a PAX block: decode either
    reports the checksum/format error or (corruption in slack bytes)
    succeeds — never crashes or reads out of bounds. The writer seals
    payload bytes with data_checksum (CRC-32C, ob_crc64_sse42
    semantics), so in-payload flips must be detected."""
    rng = np.random.default_rng(61)
    rows = 3000
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8),
                                 (abi.T_INT, 0, 19, 8),
                                 (abi.T_CHAR, 0, 0, 1)])
    vals = rng.integers(-10**6, 10**6, rows).astype(np.int64)
    small = rng.integers(0, 9, rows).astype(np.int64)
    grp = (65 + rng.integers(0, 3, rows)).astype(np.uint8)
    blk = bytearray(oracle.encode_block(
        schema, [vals.view(np.uint8), small.view(np.uint8), grp],
        [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_DICT]))
    fd = abi.make_filter([dict(col=0, op=abi.OP_GE, lo=0)])
    detected = survived = 0
    for _ in range(200):
        pos = int(rng.integers(0, len(blk)))
        old = blk[pos]
        flip = old ^ (1 << int(rng.integers(0, 8)))
        blk[pos] = flip
        try:
            oracle.filter_block(schema, 3, bytes(blk), fd)
            survived += 1     # slack/pad byte or semantic-neutral bit
        except RuntimeError:
            detected += 1
        finally:
            blk[pos] = old
    # in-payload corruption dominates; most flips must be detected
    assert detected > 150, (detected, survived)
    # sanity: the pristine block still decodes
    oracle.filter_block(schema, 3, bytes(blk), fd)


def test_cs_corruption_fuzz_never_crashes():
    """Same for a CS block through the host parser + transcode-free
    entry (obx_cs_host_parse): corrupted bytes produce clean errors,
    never crashes."""
    import ctypes as C
    from test_cs_block import _enc, _int_col
    rng = np.random.default_rng(67)
    rows = 2500
    v = rng.integers(0, 10**6, rows).astype(np.int64)
    blk = bytearray(_enc(rows, [_int_col(v, enc=5)]))
    lib = C.CDLL(os.path.join(REPO, "oceanbase_amd", "libobx.so"))
    lib.obx_cs_host_parse.restype = C.c_int
    lib.obx_cs_host_parse.argtypes = [C.c_void_p, C.c_int64,
                                      C.POINTER(C.c_uint32),
                                      C.POINTER(C.c_uint32)]
    outcomes = {"ok": 0, "err": 0}
    rows_out = C.c_uint32()
    ncols_out = C.c_uint32()
    for _ in range(200):
        pos = int(rng.integers(0, len(blk)))
        old = blk[pos]
        blk[pos] = old ^ (1 << int(rng.integers(0, 8)))
        buf = (C.c_uint8 * len(blk)).from_buffer(blk)
        rc = lib.obx_cs_host_parse(C.addressof(buf), len(blk),
                                   C.byref(rows_out), C.byref(ncols_out))
        outcomes["ok" if rc >= 0 else "err"] += 1
        blk[pos] = old
    # no crash across all 200 corruptions is the property; both outcomes
    # are legal depending on where the flip landed
    assert outcomes["ok"] + outcomes["err"] == 200, outcomes


def test_auto_encoding_roundtrip_fuzz():
    """120 random column shapes through OBX_ENC_AUTO: whatever encoding
    the writer picks, decode + filter popcount must round-trip, and the
    independent python model must agree with the C decoder."""
    import pymodel
    rng = np.random.default_rng(83)
    checked = 0
    for i in range(120):
        rows = int(rng.integers(1, 3000))
        kind = int(rng.integers(0, 6))
        if kind == 0:    # few distinct -> dict/const
            vals = rng.choice(rng.integers(-50, 50, 5), rows)
        elif kind == 1:  # runs -> rle
            vals = np.repeat(rng.integers(0, 9, max(rows // 9, 1)),
                             9)[:rows].copy()
            if len(vals) < rows:
                vals = np.pad(vals, (0, rows - len(vals)))
        elif kind == 2:  # narrow range -> intdiff/bitpack
            base = int(rng.integers(-10**9, 10**9))
            vals = base + rng.integers(0, 100, rows)
        elif kind == 3:  # constant
            vals = np.full(rows, int(rng.integers(-10**6, 10**6)))
        elif kind == 4:  # wide random -> raw
            vals = rng.integers(-10**12, 10**12, rows)
        else:            # constant + few exceptions -> const/exc
            vals = np.full(rows, 7)
            for r in range(0, rows, 97):
                vals[r] = int(rng.integers(-100, 100))
        vals = vals.astype(np.int64)
        nulls = None
        if rng.random() < 0.4:
            nulls = np.zeros((rows + 7) // 8, dtype=np.uint8)
            for r in range(0, rows, int(rng.integers(5, 40))):
                nulls[r >> 3] |= 1 << (r & 7)
        schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
        blk = oracle.encode_block(schema, [vals.view(np.uint8)],
                                  [abi.ENC_AUTO],
                                  [nulls] if nulls is not None else None)
        # python model agrees with the C encoder's bytes
        pb = pymodel.Block(blk, [(abi.T_INT, 0, 19, 8)])
        got = pb.decode_col(0)
        null_set = set()
        if nulls is not None:
            null_set = {r for r in range(rows)
                        if nulls[r >> 3] & (1 << (r & 7))}
        for r in range(rows):
            if r in null_set:
                continue
            gv = got[r][0] if isinstance(got[r], tuple) else got[r]
            assert gv == int(vals[r]), (i, r, kind)
        # filter popcount cross-check
        lo = int(np.median(vals))
        fd = abi.make_filter([dict(col=0, op=abi.OP_LE, lo=lo)])
        _bits, pc = oracle.filter_block(schema, 1, blk, fd)
        expect = sum(1 for r in range(rows)
                     if r not in null_set and int(vals[r]) <= lo)
        assert pc == expect, (i, kind)
        checked += 1
    assert checked == 120
