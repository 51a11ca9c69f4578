"""Fused scan->filter->aggregate vs an exact Python-int computation.

Covers the Q1 and Q6 shapes (BASELINE.json configs 4/5) on small generated
datasets: bit-exact group keys, counts and decimal sums (Python ints are
arbitrary precision, so the expected values are mathematically exact —
matching the reference's wide-integer accumulation, share/aggregate/sum.h).
"""
import numpy as np
import pytest

from oceanbase_amd import abi, oracle
import pymodel


def _decode_all(li):
    """Decode every block via the independent Python model -> per-column
    Python lists."""
    schema_t = [(li.schema[c].obj_type, li.schema[c].scale,
                 li.schema[c].precision, li.schema[c].len)
                for c in range(li.n_cols)]
    cols = [[] for _ in range(li.n_cols)]
    for b in range(li.n_blocks):
        pb = pymodel.Block(li.block(b), schema_t)
        for c in range(li.n_cols):
            cols[c].extend(pb.decode_col(c))
    return cols, schema_t


def q1_expected(cols):
    """cols: qty2, extprice2, disc2, tax2, returnflag, linestatus, shipdate"""
    qty, price, disc, tax, rf, ls, ship = cols
    cutoff = oracle.date_days(1998, 9, 2)
    groups = {}
    for i in range(len(qty)):
        if ship[i] > cutoff:
            continue
        key = bytes([rf[i], ls[i]])
        g = groups.setdefault(key, [0, 0, 0, 0, 0, 0, 0])
        g[0] += 1                               # count
        g[1] += qty[i]                          # sum_qty (scale 2)
        g[2] += price[i]                        # sum_base_price (scale 2)
        g[3] += price[i] * (100 - disc[i])      # sum_disc_price (scale 4)
        g[4] += price[i] * (100 - disc[i]) * (100 + tax[i])  # charge (scale 6)
        g[5] += disc[i]                         # sum_disc (scale 2)
    return dict(sorted(groups.items()))


def q1_descs():
    filt = abi.make_filter([
        dict(col=6, op=abi.OP_LE, lo=oracle.date_days(1998, 9, 2))])
    agg = abi.make_agg([4, 5], [
        dict(kind=abi.AGG_COUNT),
        dict(kind=abi.AGG_SUM, col_a=0),
        dict(kind=abi.AGG_SUM, col_a=1),
        dict(kind=abi.AGG_SUM_PROD2, col_a=1, col_b=2),
        dict(kind=abi.AGG_SUM_PROD3, col_a=1, col_b=2, col_c=3),
        dict(kind=abi.AGG_SUM, col_a=2),
    ])
    return filt, agg


def _check_q1(res, cols):
    expect = q1_expected(cols)
    rows = abi.result_rows(res, 6)
    assert len(rows) == len(expect)
    for (key, rc, cells), (ekey, eg) in zip(rows, expect.items()):
        assert key == ekey
        assert rc == eg[0]
        assert cells[0] == eg[0]      # COUNT
        assert cells[1] == eg[1]      # sum qty
        assert cells[2] == eg[2]      # sum price
        assert cells[3] == eg[3]      # sum disc_price
        assert cells[4] == eg[4]      # sum charge
        assert cells[5] == eg[5]      # sum disc


def test_q1_small():
    li = oracle.Lineitem(4, 20000, seed=42, block_bytes=4096)
    assert li.n_blocks > 10
    cols, _ = _decode_all(li)
    filt, agg = q1_descs()
    res = oracle.scan_filter_agg(li.bs, filt, agg, nthreads=2)
    assert res.rows_scanned == 20000
    _check_q1(res, cols)


def test_q1_thread_count_invariant():
    li = oracle.Lineitem(4, 8000, seed=1, block_bytes=4096)
    filt, agg = q1_descs()
    r1 = oracle.scan_filter_agg(li.bs, filt, agg, nthreads=1)
    r8 = oracle.scan_filter_agg(li.bs, filt, agg, nthreads=8)
    assert abi.result_rows(r1, 6) == abi.result_rows(r8, 6)


def test_q6_small():
    li = oracle.Lineitem(6, 30000, seed=42, block_bytes=4096)
    cols, _ = _decode_all(li)
    ship, disc, qty, price = cols
    d94, d95 = oracle.date_days(1994, 1, 1), oracle.date_days(1995, 1, 1)
    expect_rev = 0
    expect_cnt = 0
    for i in range(len(ship)):
        if d94 <= ship[i] < d95 and 5 <= disc[i] <= 7 and qty[i] < 2400:
            expect_rev += price[i] * disc[i]
            expect_cnt += 1
    filt = abi.make_filter([
        dict(col=0, op=abi.OP_GE, lo=d94),
        dict(col=0, op=abi.OP_LT, lo=d95),
        dict(col=1, op=abi.OP_BT, lo=5, hi=7),
        dict(col=2, op=abi.OP_LT, lo=2400)])
    agg = abi.make_agg([], [dict(kind=abi.AGG_SUM_MUL, col_a=3, col_b=1)])
    res = oracle.scan_filter_agg(li.bs, filt, agg)
    assert res.n_groups == 1
    assert res.rows_passed == expect_cnt
    rows = abi.result_rows(res, 1)
    assert rows[0][1] == expect_cnt
    assert rows[0][2][0] == expect_rev


def test_config2_count_filter():
    li = oracle.Lineitem(2, 50000, seed=42)
    cols, _ = _decode_all(li)
    qty = cols[0]
    expect = sum(1 for v in qty if v < 24)
    filt = abi.make_filter([dict(col=0, op=abi.OP_LT, lo=24)])
    res = oracle.scan_filter_agg(li.bs, filt, None)
    assert res.rows_scanned == 50000
    assert res.rows_passed == expect
    # ~46% selectivity sanity (uniform 1..50 -> 23/50)
    assert 0.4 < expect / 50000 < 0.52


def test_config3_decode_filter():
    li = oracle.Lineitem(3, 20000, seed=9)
    cols, schema_t = _decode_all(li)
    ship = cols[0]
    cutoff = oracle.date_days(1998, 9, 2)
    expect = sum(1 for v in ship if v <= cutoff)
    filt = abi.make_filter([dict(col=0, op=abi.OP_LE, lo=cutoff)])
    res = oracle.scan_filter_agg(li.bs, filt, None)
    assert res.rows_passed == expect
    # encodings as configured
    pb = pymodel.Block(li.block(0), schema_t)
    types = [h["type"] for h in pb.col_headers]
    assert types == [pymodel.ENC_INT_DIFF, pymodel.ENC_RAW, pymodel.ENC_RLE,
                     pymodel.ENC_DICT]


def test_min_max_aggs():
    li = oracle.Lineitem(4, 5000, seed=3, block_bytes=4096)
    cols, _ = _decode_all(li)
    agg = abi.make_agg([], [dict(kind=abi.AGG_MIN, col_a=1),
                            dict(kind=abi.AGG_MAX, col_a=1),
                            dict(kind=abi.AGG_COUNT)])
    res = oracle.scan_filter_agg(li.bs, None, agg)
    rows = abi.result_rows(res, 3)
    assert rows[0][2][0] == min(cols[1])
    assert rows[0][2][1] == max(cols[1])
    assert rows[0][2][2] == 5000


def test_generator_shard_determinism():
    """Sharding by row ranges reproduces identical bytes (multi-GPU
    contract: every rank generates its own shard)."""
    full = oracle.Lineitem(4, 12000, seed=42, block_bytes=4096)
    rpb = None
    # infer rows per block from offsets/row counts
    b0 = full.block(0)
    rpb = int(np.frombuffer(b0[16:20], dtype=np.uint32)[0])
    cut_blocks = full.n_blocks // 2
    cut_rows = cut_blocks * rpb
    a = oracle.Lineitem(4, cut_rows, seed=42, block_bytes=4096, row_base=0)
    b = oracle.Lineitem(4, 12000 - cut_rows, seed=42, block_bytes=4096,
                        row_base=cut_rows)
    assert a.n_blocks + b.n_blocks == full.n_blocks
    for i in range(a.n_blocks):
        assert a.block(i) == full.block(i)
    for i in range(b.n_blocks):
        assert b.block(i) == full.block(cut_blocks + i)
