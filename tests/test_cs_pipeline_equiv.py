"""Cross-format hot-path equivalence: the same logical table stored as
PAX micro-blocks (the north-star path's format) and as CS
(cs_encoding) micro-blocks must yield identical filter+aggregate
results — the PAX side through the oracle's scan_filter_agg pipeline,
the CS side through the CS block decoder (obx_cs_block_get_int) with
the same white-filter/aggregate semantics applied to the decoded
columns. This ties SURVEY §8(f) row 2's format restatement back to the
hot path's observable results."""
import ctypes as C
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oceanbase_amd import abi, oracle  # noqa: E402

from test_cs_block import (  # noqa: E402
    BlockView, _enc as cs_enc, _dec as cs_dec, _get_int as cs_get_int,
    _int_col as cs_int_col,
)


def _pax_blockset(schema, blocks):
    import ctypes as Ct
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


@pytest.mark.parametrize("seed", [0, 1, 2])
def test_q6_shape_equivalence(seed):
    """Q6-style scan (range filter + grand-total SUMs) over the same
    rows in both formats."""
    rng = np.random.default_rng(500 + seed)
    rows_total = 12000
    rpb = 1500
    qty = rng.integers(1, 51, rows_total).astype(np.int64)
    price = rng.integers(900, 105000, rows_total).astype(np.int64)
    ship = rng.integers(8000, 11000, rows_total).astype(np.int64)
    null_rows = set(int(x) for x in rng.choice(rows_total, 300,
                                               replace=False))
    nb_full = np.zeros((rows_total + 7) // 8, dtype=np.uint8)
    for r in null_rows:
        nb_full[r >> 3] |= 1 << (r & 7)

    # ---- PAX side: oracle pipeline ----
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)] * 3)
    blocks = []
    for r0 in range(0, rows_total, rpb):
        n = min(rpb, rows_total - r0)
        nb = np.zeros((n + 7) // 8, dtype=np.uint8)
        for r in range(n):
            if (r0 + r) in null_rows:
                nb[r >> 3] |= 1 << (r & 7)
        blocks.append(oracle.encode_block(
            schema,
            [qty[r0:r0 + n].copy(), price[r0:r0 + n].copy(),
             ship[r0:r0 + n].copy()],
            [abi.ENC_AUTO] * 3, [None, None, nb]))
    bs = _pax_blockset(schema, blocks)
    cutoff = 9500
    filt = abi.make_filter([dict(col=2, op=abi.OP_LT, lo=cutoff)])
    agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT),
                            dict(kind=abi.AGG_SUM, col_a=0),
                            dict(kind=abi.AGG_SUM, col_a=1)])
    res = oracle.scan_filter_agg(bs, filt, agg)
    pax_rows = abi.result_rows(res, 3)

    # ---- CS side: format decode + the same semantics ----
    got_qty = np.zeros(rows_total, dtype=np.int64)
    got_price = np.zeros(rows_total, dtype=np.int64)
    got_ship = np.zeros(rows_total, dtype=np.int64)
    got_null = np.zeros(rows_total, dtype=bool)
    encs = [0, 6, 2]  # RAW / SIMD_FIXEDPFOR / DDZR per column
    for r0 in range(0, rows_total, rpb):
        n = min(rpb, rows_total - r0)
        nulls_blk = [r for r in range(n) if (r0 + r) in null_rows]
        cols = [
            cs_int_col(list(qty[r0:r0 + n]), enc=encs[0]),
            cs_int_col(list(price[r0:r0 + n]), enc=encs[1]),
            cs_int_col(list(ship[r0:r0 + n]), enc=encs[2],
                       null_rows=nulls_blk or None),
        ]
        v = cs_dec(cs_enc(n, cols))
        for arr, c in ((got_qty, 0), (got_price, 1), (got_ship, 2)):
            out, nset = cs_get_int(v, c)
            arr[r0:r0 + n] = out
            if c == 2:
                for r in nset:
                    got_null[r0 + r] = True
    assert got_null.sum() == len(null_rows)
    # white-filter semantics: NULL fails the comparison
    mask = (~got_null) & (got_ship < cutoff)
    cs_count = int(mask.sum())
    cs_sum_qty = int(got_qty[mask].sum())
    cs_sum_price = int(got_price[mask].sum())

    assert res.rows_passed == cs_count
    assert len(pax_rows) == 1
    _key, row_count, cells = pax_rows[0]
    assert row_count == cs_count
    assert cells[0] == cs_count
    assert cells[1] == cs_sum_qty
    assert cells[2] == cs_sum_price


def test_groupby_equivalence_with_dict_column():
    """Q1-style group-by over a low-cardinality column stored as
    INT_DICT on the CS side."""
    rng = np.random.default_rng(77)
    rows = 9000
    flag = rng.integers(0, 3, rows).astype(np.int64)   # group col
    qty = rng.integers(1, 50, rows).astype(np.int64)

    schema = oracle.make_schema([(abi.T_INT, 0, 19, 1),
                                 (abi.T_INT, 0, 19, 8)])
    # group col must be byte-width for the key: store as 1-byte column
    flag_b = flag.astype(np.uint8)
    blocks = [oracle.encode_block(
        schema, [flag_b[r0:r0 + 3000].copy(), qty[r0:r0 + 3000].copy()],
        [abi.ENC_AUTO, abi.ENC_AUTO], None)
        for r0 in range(0, rows, 3000)]
    bs = _pax_blockset(schema, blocks)
    agg = abi.make_agg([0], [dict(kind=abi.AGG_COUNT),
                             dict(kind=abi.AGG_SUM, col_a=1)])
    res = oracle.scan_filter_agg(bs, None, agg)
    pax = {k: (rc, cells) for k, rc, cells in abi.result_rows(res, 2)}

    # CS side with the flag column dictionary-encoded
    got_flag = np.zeros(rows, dtype=np.int64)
    got_qty = np.zeros(rows, dtype=np.int64)
    for r0 in range(0, rows, 3000):
        v = cs_dec(cs_enc(3000, [
            cs_int_col(list(flag[r0:r0 + 3000]), dict_=True),
            cs_int_col(list(qty[r0:r0 + 3000]), enc=5),
        ]))
        got_flag[r0:r0 + 3000] = cs_get_int(v, 0)[0]
        got_qty[r0:r0 + 3000] = cs_get_int(v, 1)[0]

    assert len(pax) == 3
    for g in range(3):
        m = got_flag == g
        key = bytes([g])
        assert key in pax
        rc, cells = pax[key]
        assert rc == int(m.sum())
        assert cells[0] == int(m.sum())
        assert cells[1] == int(got_qty[m].sum())
