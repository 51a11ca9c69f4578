"""Randomized GPU/oracle parity fuzzing.

Deterministic (seeded) random schemas x encodings x NULL densities x white
filters x aggregate plans, compared bit-exactly between the HIP engine and
the CPU oracle. Mirrors the reference's randomized row-pattern round-trip
tests (ObRowGenerate in test_column_decoder.h) but adversarially across the
whole plan space the engine supports.
"""
import ctypes as Ct
import os

import numpy as np
import pytest

from oceanbase_amd import abi, oracle

pytestmark = pytest.mark.gpu

TYPES = [
    (abi.T_INT, 0, 19, 8),
    (abi.T_INT32, 0, 9, 4),
    (abi.T_DATE, 0, 0, 4),
    (abi.T_DECIMAL_INT, 2, 15, 8),
    (abi.T_CHAR, 0, 0, 1),
    (abi.T_CHAR, 0, 0, 4),
]
ENCS = [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_RLE, abi.ENC_INT_DIFF,
        abi.ENC_AUTO]


def _gen_column(rng, tspec, rows, style):
    t, scale, prec, length = tspec
    if t == abi.T_CHAR:
        alpha = np.frombuffer(b"ABCDNRXYZ", dtype=np.uint8)
        if length == 1:
            vals = rng.choice(alpha[: rng.integers(2, 9)], rows)
            return vals, np.int64(vals[:]), length
        # multi-byte chars: common-prefix + small-alphabet suffix shapes
        # (exercises DICT/STRING_DIFF/HEX_PACKING/STRING_PREFIX writers)
        a2 = np.empty((rows, length), dtype=np.uint8)
        n_pref = int(rng.integers(1, 4))
        prefs = [bytes(rng.choice(alpha, length - 1)) for _ in range(n_pref)]
        pick = rng.integers(0, n_pref, rows)
        for r in range(rows):
            a2[r, :length - 1] = np.frombuffer(prefs[pick[r]],
                                               dtype=np.uint8)
        a2[:, length - 1] = rng.choice(alpha[:4], rows)
        v = np.array([int.from_bytes(a2[r].tobytes(), "little")
                      for r in range(rows)], dtype=np.int64)
        return a2.reshape(-1), v, length
    if style == 0:      # tiny domain (dict/rle friendly)
        dom = rng.integers(-50, 50, rng.integers(2, 12)).astype(np.int64)
        v = rng.choice(dom, rows)
    elif style == 1:    # runs
        v = np.repeat(rng.integers(-100, 100, rows // 50 + 1), 50)[:rows]
        v = v.astype(np.int64)
    elif style == 2:    # narrow range (int-diff friendly)
        base = int(rng.integers(-10**6, 10**6))
        v = (base + rng.integers(0, 4000, rows)).astype(np.int64)
    else:               # wide
        lim = 2**30 if length == 4 else 2**60
        v = rng.integers(-lim, lim, rows).astype(np.int64)
    if length == 4:
        v = np.clip(v, -2**31, 2**31 - 1)
        arr = v.astype(np.int32).view(np.uint8)
    else:
        arr = v.view(np.uint8)
    return arr, v, length


def _encoding_for(rng, tspec, enc, style):
    t = tspec[0]
    if enc == abi.ENC_INT_DIFF and t == abi.T_CHAR:
        # chars can't int-diff: exercise the string transforms instead
        if tspec[3] > 1 and style == 3:
            return abi.ENC_STRING_PREFIX
        return abi.ENC_HEX if style % 2 else abi.ENC_SDIFF
    if enc == abi.ENC_INT_DIFF and t == abi.T_DECIMAL_INT:
        return abi.ENC_RAW
    if enc == abi.ENC_RLE and t == abi.T_CHAR:
        return abi.ENC_DICT
    # dict/rle on wide-random data is a writer the reference's cost model
    # would never choose and blows up meta sizes — keep them to the small
    # domains (the oracle still round-trips them; not parity-relevant)
    if t != abi.T_CHAR and enc in (abi.ENC_DICT, abi.ENC_RLE) and style > 1:
        return abi.ENC_RAW
    return enc


def _blockset(schema, blocks):
    aligned, offs = [], [0]
    for b in blocks:
        body = b[:-16]
        pad = (-len(body)) % 16
        aligned.append(body + b"\x00" * pad)
        offs.append(offs[-1] + len(body) + pad)
    data = np.frombuffer(b"".join(aligned) + b"\x00" * 16, dtype=np.uint8)
    offarr = np.array(offs, dtype=np.uint64)
    bs = abi.BlockSet()
    bs.data = data.ctypes.data_as(Ct.POINTER(Ct.c_uint8))
    bs.block_offsets = offarr.ctypes.data_as(Ct.POINTER(Ct.c_uint64))
    bs.n_blocks = len(blocks)
    bs.n_cols = len(schema)
    bs.cols = Ct.cast(schema, Ct.POINTER(abi.ColSchema))
    bs._keep = (data, offarr)
    return bs


@pytest.mark.parametrize("case_seed",
                         range(int(os.environ.get("OBX_FUZZ_SEEDS", "30"))))
def test_fuzz_case(case_seed):
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(1000 + case_seed)
    n_cols = int(rng.integers(1, 6))
    tspecs = [TYPES[rng.integers(0, len(TYPES))] for _ in range(n_cols)]
    styles = [int(rng.integers(0, 4)) for _ in range(n_cols)]
    encs = [_encoding_for(rng, tspecs[c],
                          ENCS[rng.integers(0, len(ENCS))], styles[c])
            for c in range(n_cols)]
    rows_total = int(rng.integers(3000, 40000))
    rows_per_block = int(rng.integers(200, 3000))
    schema = oracle.make_schema(tspecs)

    blocks = []
    distinct = [set() for _ in range(n_cols)]
    coleq_cols = set()
    r0 = 0
    while r0 < rows_total:
        n = min(rows_per_block, rows_total - r0)
        arrays, nulls = [], []
        for c in range(n_cols):
            if c in coleq_cols:
                prev = next(j for j in reversed(range(c))
                            if tspecs[j][3] == tspecs[c][3]
                            and _sc(tspecs[j]) == _sc(tspecs[c])
                            and j not in coleq_cols)
                arr = arrays[prev].copy()
                ln = tspecs[c][3]
                for r in rng.choice(n, max(1, n // 50), replace=False):
                    arr[r * ln] ^= 0x5
                arrays.append(arr)
                nulls.append(nulls[prev])
                continue
            arr, v, length = _gen_column(rng, tspecs[c], n, styles[c])
            if len(distinct[c]) <= 40:
                distinct[c].update(np.unique(v)[:41].tolist())
            nb = None
            if rng.random() < 0.3 and encs[c] != abi.ENC_RLE:
                nb = np.zeros((n + 7) // 8, dtype=np.uint8)
                for r in rng.choice(n, max(1, n // 17), replace=False):
                    nb[r >> 3] |= 1 << (r & 7)
            arrays.append(arr)
            nulls.append(nb)
        blocks.append(oracle.encode_block(schema, arrays, encs, nulls))
        r0 += n
    bs = _blockset(schema, blocks)

    # random AND filter over 0..3 leaves
    leaves = []
    for _ in range(int(rng.integers(0, 4))):
        c = int(rng.integers(0, n_cols))
        op = int(rng.integers(0, 10))
        lo = int(rng.integers(-1000, 1000))
        hi = lo + int(rng.integers(0, 2000))
        leaf = dict(col=c, op=op, lo=lo, hi=hi)
        if op == abi.OP_IN:
            leaf["in_list"] = [int(x) for x in
                               rng.integers(-100, 100, rng.integers(1, 6))]
        leaves.append(leaf)
    filt = abi.make_filter(leaves)

    # random aggregate plan: group by <=2 byte-compatible cols, <=5 aggs
    gcands = [c for c in range(n_cols)
              if len(distinct[c]) <= 10 and c not in coleq_cols]
    rng.shuffle(gcands)
    group = []
    klen = 0
    card = 1
    for c in gcands[:2]:
        d = max(len(distinct[c]), 1) + 1  # + null bucket
        if klen + tspecs[c][3] <= 7 and card * d <= 40 and rng.random() < 0.7:
            group.append(c)
            klen += tspecs[c][3]
            card *= d
    aggs = [dict(kind=abi.AGG_COUNT)]
    for _ in range(int(rng.integers(0, 4))):
        kind = int(rng.choice([abi.AGG_SUM, abi.AGG_MIN, abi.AGG_MAX,
                               abi.AGG_COUNT]))
        aggs.append(dict(kind=kind, col_a=int(rng.integers(0, n_cols))))
    agg = abi.make_agg(group, aggs)

    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    if res_cpu.n_groups > 60:
        pytest.skip("group cardinality beyond round-1 table bound")
    eng = GpuEngine(0)
    h = eng.load(bs)
    try:
        res_gpu = eng.scan_filter_agg(h, filt, agg)
    except RuntimeError as e:
        if "-4009" in str(e):  # >254 live groups: documented bound
            pytest.skip("group overflow (documented round-1 bound)")
        raise
    assert res_gpu.rows_passed == res_cpu.rows_passed, case_seed
    assert abi.result_rows(res_gpu, len(aggs)) == \
        abi.result_rows(res_cpu, len(aggs)), case_seed
    # filter-only path too
    survivors = eng.filter(h, filt)
    assert survivors == res_cpu.rows_passed
    eng.close()
