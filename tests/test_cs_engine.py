"""CS blocks through the engine hot path via the transformer-role load
(oceanbase_amd/cs.py): CS disk format -> load-time transcode -> the
SAME PAX scan kernels. Mirrors ObCSMicroBlockTransformer's position in
the reference stack (transform on load, vectorized scan after)."""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oceanbase_amd import abi, oracle  # noqa: E402
import cs_oracle_util as cs  # noqa: E402

from test_cs_block import (  # noqa: E402
    _enc as cs_enc, _int_col as cs_int_col, _str_col as cs_str_col,
)
from test_cs_pipeline_equiv import _pax_blockset  # noqa: E402


def _make_cs_table(seed=11, rows_total=8000, rpb=2000):
    rng = np.random.default_rng(seed)
    flag = rng.integers(0, 3, rows_total).astype(np.int64)
    qty = rng.integers(1, 50, rows_total).astype(np.int64)
    ship = rng.integers(8000, 11000, rows_total).astype(np.int64)
    tags = [bytes("T%d" % (f % 3), "ascii") for f in flag]
    null_rows = set(int(x) for x in rng.choice(rows_total, 200,
                                               replace=False))
    blocks = []
    for r0 in range(0, rows_total, rpb):
        n = min(rpb, rows_total - r0)
        nblk = [r for r in range(n) if (r0 + r) in null_rows]
        blocks.append(cs_enc(n, [
            cs_int_col(list(flag[r0:r0 + n]), dict_=True),
            cs_int_col(list(qty[r0:r0 + n]), enc=6),
            cs_int_col(list(ship[r0:r0 + n]), enc=2,
                       null_rows=nblk or None),
            cs_str_col(tags[r0:r0 + n], dict_=True),
        ]))
    return blocks, flag, qty, ship, tags, null_rows


def _expected(flag, qty, ship, null_rows, cutoff):
    nul = np.zeros(len(ship), dtype=bool)
    for r in null_rows:
        nul[r] = True
    mask = (~nul) & (ship < cutoff)
    exp = {}
    for g in range(3):
        m = mask & (flag == g)
        exp[g] = (int(m.sum()), int(qty[m].sum()))
    return exp


def _plan(cutoff):
    filt = abi.make_filter([dict(col=2, op=abi.OP_LT, lo=cutoff)])
    agg = abi.make_agg([0], [dict(kind=abi.AGG_COUNT),
                             dict(kind=abi.AGG_SUM, col_a=1)])
    return filt, agg


def test_cs_load_cpu_pipeline():
    blocks, flag, qty, ship, tags, null_rows = _make_cs_table()
    schema, pax = cs.to_pax_blocks(blocks)
    assert [(s.obj_type, s.len) for s in schema] == \
        [(abi.T_INT, 8)] * 3 + [(abi.T_CHAR, 2)]
    bs = _pax_blockset(schema, pax)
    cutoff = 9500
    filt, agg = _plan(cutoff)
    res = oracle.scan_filter_agg(bs, filt, agg)
    exp = _expected(flag, qty, ship, null_rows, cutoff)
    got = {}
    for key, rc, cells in abi.result_rows(res, 2):
        g = int(np.frombuffer(key.ljust(8, b"\x00"),
                              dtype=np.int64)[0])
        got[g] = (rc, cells[1])
    for g in range(3):
        assert got[g][0] == exp[g][0] == \
            [c for k, r, c in abi.result_rows(res, 2)
             if int(np.frombuffer(k.ljust(8, b"\x00"),
                                  dtype=np.int64)[0]) == g][0][0]
        assert got[g][1] == exp[g][1]


def test_cs_var_strings_rejected():
    rng = np.random.default_rng(5)
    strs = [b"v" * int(rng.integers(1, 9)) for _ in range(100)]
    blocks = [cs_enc(100, [cs_str_col(strs)])]
    with pytest.raises(ValueError):
        cs.to_pax_blocks(blocks)


@pytest.mark.gpu
def test_cs_load_gpu_scan_parity():
    """CS-encoded table scanned by the GPU engine after the load-time
    transform; results must match the CPU oracle on the same blocks."""
    from oceanbase_amd.engine import GpuEngine
    blocks, flag, qty, ship, tags, null_rows = _make_cs_table(seed=23)
    schema, pax = cs.to_pax_blocks(blocks)
    bs = _pax_blockset(schema, pax)
    cutoff = 9300
    filt, agg = _plan(cutoff)
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    eng = GpuEngine(0)
    h = eng.load(bs)
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    assert abi.result_rows(res_gpu, 2) == abi.result_rows(res_cpu, 2)
    exp = _expected(flag, qty, ship, null_rows, cutoff)
    assert res_cpu.rows_passed == sum(v[0] for v in exp.values())
    eng.close()


def test_cs_load_preserves_nullness_for_nu_nn():
    """IS NULL / IS NOT NULL white filters over CS-loaded columns: the
    transcode must preserve nullness exactly for every CS null
    representation (replace-value, bitmap, dict null ref, zero-len
    string)."""
    rows = 3000
    rng = np.random.default_rng(101)
    vals_pos = rng.integers(10, 10**6, rows).astype(np.int64)   # replace=min-1
    vals_full = rng.integers(-2**63, 2**63 - 1, rows).astype(np.int64)
    vals_full[0], vals_full[1] = -2**63, 2**63 - 1              # bitmap
    flag = rng.integers(0, 4, rows).astype(np.int64)            # dict null ref
    tags = [bytes("N%d" % (f % 4), "ascii") for f in flag]
    nulls = [sorted(int(x) for x in rng.choice(rows, k, replace=False))
             for k in (100, 150, 200, 250)]
    blocks = [cs_enc(rows, [
        cs_int_col(list(vals_pos), null_rows=nulls[0]),
        cs_int_col(list(vals_full), null_rows=nulls[1]),
        cs_int_col(list(flag), null_rows=nulls[2], dict_=True),
        cs_str_col(tags, null_rows=nulls[3], dict_=True),
    ])]
    schema, pax = cs.to_pax_blocks(blocks)
    bs = _pax_blockset(schema, pax)
    for c in range(4):
        for op, expect in ((abi.OP_NU, len(nulls[c])),
                           (abi.OP_NN, rows - len(nulls[c]))):
            filt = abi.make_filter([dict(col=c, op=op)])
            agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT)])
            res = oracle.scan_filter_agg(bs, filt, agg)
            assert res.rows_passed == expect, (c, op)
