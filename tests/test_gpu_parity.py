"""GPU parity: the HIP engine vs the CPU oracle on identical microblock bytes.

Bit-exact contract (BASELINE.json north_star): integer columns, COUNT,
group keys bit-exact; decimal sums here are exact integers, so they are
bit-exact too. The oracle is the checker, never the measured path.
"""
import numpy as np
import pytest

from oceanbase_amd import abi, oracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from oceanbase_amd.engine import GpuEngine
    e = GpuEngine(0)
    yield e
    e.close()


def _load_lineitem(eng, config, rows, seed=42, block_bytes=16384):
    li = oracle.Lineitem(config, rows, seed=seed, block_bytes=block_bytes)
    h = eng.load(li.bs)
    return li, h


def test_decode_parity_all_encodings(eng):
    """k_decode vs oracle decode on every config-4 column (dict/raw/intdiff)
    + config-3 (rle, raw bitpack)."""
    for config, rows in ((4, 30000), (3, 30000), (2, 30000)):
        li, h = _load_lineitem(eng, config, rows, block_bytes=8192)
        cols = list(range(li.n_cols))
        eng.decode(h, cols)
        # oracle decode block by block
        schema = li.schema
        for c in cols:
            gpu = eng.fetch_col(h, c, schema[c].len)
            off = 0
            for b in range(li.n_blocks):
                blk = li.block(b)
                rcount, outs, _ = oracle.decode_block(schema, li.n_cols, blk,
                                                      [c])
                want = outs[0]
                got = gpu[off * schema[c].len:(off + rcount) * schema[c].len]
                assert np.array_equal(got, want), (config, c, b)
                off += rcount
        eng.free(h)


def test_filter_parity_config2(eng):
    li, h = _load_lineitem(eng, 2, 200000)
    fd = abi.make_filter([dict(col=0, op=abi.OP_LT, lo=24)])
    survivors = eng.filter(h, fd, want_row_ids=True)
    bm = eng.fetch_bitmap(h)
    row_ids, n = eng.fetch_row_ids(h)
    blk_counts = eng.fetch_blk_counts(h, li.n_blocks)
    # oracle bitmap, block by block
    off = 0
    total = 0
    for b in range(li.n_blocks):
        blk = li.block(b)
        bits, pc = oracle.filter_block(li.schema, li.n_cols, blk, fd)
        total += pc
        assert blk_counts[b] == pc, b
        # compare bit ranges
        for r in range(len(bits) * 8):
            g = off + r
            if g >= off + int(np.frombuffer(blk[16:20], dtype=np.uint32)[0]):
                break
            want = (bits[r >> 3] >> (r & 7)) & 1
            got = (bm[g >> 3] >> (g & 7)) & 1
            assert got == want, (b, r)
        # row_ids: survivors' block-local ids, ascending
        rows_b = int(np.frombuffer(blk[16:20], dtype=np.uint32)[0])
        want_ids = [r for r in range(rows_b) if (bits[r >> 3] >> (r & 7)) & 1]
        got_ids = row_ids[off:off + pc].tolist()
        assert got_ids == want_ids, b
        off += rows_b
    assert survivors == total


def test_filter_parity_config3(eng):
    li, h = _load_lineitem(eng, 3, 100000)
    cutoff = oracle.date_days(1998, 9, 2)
    fd = abi.make_filter([dict(col=0, op=abi.OP_LE, lo=cutoff),
                          dict(col=2, op=abi.OP_EQ, lo=ord("F")),
                          dict(col=3, op=abi.OP_BT, lo=3, hi=8)])
    survivors = eng.filter(h, fd, want_row_ids=False)
    res = oracle.scan_filter_agg(li.bs, fd, None)
    assert survivors == res.rows_passed


@pytest.mark.parametrize("op,lo,hi", [
    (abi.OP_EQ, 24, 0), (abi.OP_NE, 24, 0), (abi.OP_LT, 1, 0),
    (abi.OP_LE, 50, 0), (abi.OP_GT, 49, 0), (abi.OP_GE, 51, 0),
    (abi.OP_BT, 10, 20), (abi.OP_IN, 0, 0),
])
def test_filter_ops_parity(eng, op, lo, hi):
    li, h = _load_lineitem(eng, 2, 50000)
    leaf = dict(col=0, op=op, lo=lo, hi=hi)
    if op == abi.OP_IN:
        leaf["in_list"] = [1, 24, 50]
    fd = abi.make_filter([leaf])
    survivors = eng.filter(h, fd)
    res = oracle.scan_filter_agg(li.bs, fd, None)
    assert survivors == res.rows_passed, (op, lo, hi)
    eng.free(h)


def _q1_descs():
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


def test_q1_agg_parity(eng):
    li, h = _load_lineitem(eng, 4, 200000)
    filt, agg = _q1_descs()
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(li.bs, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    rows_gpu = abi.result_rows(res_gpu, 6)
    rows_cpu = abi.result_rows(res_cpu, 6)
    assert rows_gpu == rows_cpu  # keys, counts, exact 256-bit sums
    eng.free(h)


def test_q6_agg_parity(eng):
    li, h = _load_lineitem(eng, 6, 150000)
    d94, d95 = oracle.date_days(1994, 1, 1), oracle.date_days(1995, 1, 1)
    filt = abi.make_filter([
        dict(col=0, op=abi.OP_GE, lo=d94),
        dict(col=0, op=abi.OP_LT, lo=d95),
        dict(col=1, op=abi.OP_BT, lo=5, hi=7),
        dict(col=2, op=abi.OP_LT, lo=2400)])
    agg = abi.make_agg([], [dict(kind=abi.AGG_SUM_MUL, col_a=3, col_b=1)])
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(li.bs, filt, agg)
    assert abi.result_rows(res_gpu, 1) == abi.result_rows(res_cpu, 1)
    eng.free(h)


def test_agg_with_nulls_parity(eng):
    """Hand-built blockset with NULLs in a dict column + min/max aggs."""
    rng = np.random.default_rng(5)
    rows = 5000
    vals = rng.integers(0, 9, rows).astype(np.int64)
    nulls = np.zeros((rows + 7) // 8, dtype=np.uint8)
    for r in range(0, rows, 13):
        nulls[r >> 3] |= 1 << (r & 7)
    grp = rng.choice(np.frombuffer(b"XY", dtype=np.uint8), rows)
    schema = oracle.make_schema([(abi.T_DECIMAL_INT, 2, 15, 8),
                                 (abi.T_CHAR, 0, 0, 1)])
    block = oracle.encode_block(schema, [vals.view(np.uint8), grp],
                                [abi.ENC_RAW, abi.ENC_DICT], [nulls, None])
    import ctypes as Ct
    data = np.frombuffer(block, dtype=np.uint8)
    offs = np.array([0, len(block) - 16], dtype=np.uint64)
    bs = abi.BlockSet()
    bs.data = data.ctypes.data_as(Ct.POINTER(Ct.c_uint8))
    bs.block_offsets = offs.ctypes.data_as(Ct.POINTER(Ct.c_uint64))
    bs.n_blocks = 1
    bs.n_cols = 2
    bs.cols = Ct.cast(schema, Ct.POINTER(abi.ColSchema))
    bs.total_rows = rows
    agg = abi.make_agg([1], [dict(kind=abi.AGG_SUM, col_a=0),
                             dict(kind=abi.AGG_COUNT, col_a=0),
                             dict(kind=abi.AGG_MIN, col_a=0),
                             dict(kind=abi.AGG_MAX, col_a=0),
                             dict(kind=abi.AGG_COUNT)])
    res_cpu = oracle.scan_filter_agg(bs, None, agg)
    from oceanbase_amd.engine import GpuEngine
    h = eng.load(bs)
    res_gpu = eng.scan_filter_agg(h, None, agg)
    assert abi.result_rows(res_gpu, 5) == abi.result_rows(res_cpu, 5)
    eng.free(h)
