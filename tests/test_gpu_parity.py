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


def test_row_ids_through_filter_jit(eng):
    """Selection vectors (get_row_ids shape) through the staged filter
    JIT's mask-capture path: ordered block-local row ids + per-block
    counts, bit-exact vs the oracle, with the JIT engaged."""
    li, h = _load_lineitem(eng, 3, 120000)
    cutoff = oracle.date_days(1997, 6, 1)
    fd = abi.make_filter([dict(col=0, op=abi.OP_LE, lo=cutoff)])
    survivors = eng.filter(h, fd, want_row_ids=True)
    assert eng._lib.obx_gpu_last_jit(eng._ctx) == 2  # RID JIT engaged
    row_ids, _n = eng.fetch_row_ids(h)
    blk_counts = eng.fetch_blk_counts(h, li.n_blocks)
    off = 0
    total = 0
    for b in range(li.n_blocks):
        blk = li.block(b)
        bits, pc = oracle.filter_block(li.schema, li.n_cols, blk, fd)
        total += pc
        assert blk_counts[b] == pc, b
        rows_b = int(np.frombuffer(blk[16:20], dtype=np.uint32)[0])
        want_ids = [r for r in range(rows_b)
                    if (bits[r >> 3] >> (r & 7)) & 1]
        got_ids = row_ids[off:off + pc].tolist()
        assert got_ids == want_ids, b
        off += rows_b
    assert survivors == total


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


def _manual_blockset(schema, blocks_bytes):
    """Build a BlockSet over concatenated (16-B aligned) encoded blocks."""
    import ctypes as Ct
    aligned = []
    offs = [0]
    for b in blocks_bytes:
        body = b[:-16]  # strip the slack suffix encode_block appends
        pad = (-len(body)) % 16
        aligned.append(body + b"\x00" * pad)
        offs.append(offs[-1] + len(body) + pad)
    data = np.frombuffer(b"".join(aligned) + b"\x00" * 16, dtype=np.uint8)
    offarr = np.array(offs, dtype=np.uint64)
    bs = abi.BlockSet()
    bs.data = data.ctypes.data_as(Ct.POINTER(Ct.c_uint8))
    bs.block_offsets = offarr.ctypes.data_as(Ct.POINTER(Ct.c_uint64))
    bs.n_blocks = len(blocks_bytes)
    bs.n_cols = len(schema)
    bs.cols = Ct.cast(schema, Ct.POINTER(abi.ColSchema))
    bs._keep = (data, offarr)  # keep alive
    return bs


def test_generic_path_large_blocks(eng):
    """Blocks > LDS stage size take the non-LDS kernels; parity must hold."""
    li = oracle.Lineitem(4, 120000, seed=11, block_bytes=65536)
    h = eng.load(li.bs)
    filt, agg = _q1_descs()
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(li.bs, filt, agg)
    assert abi.result_rows(res_gpu, 6) == abi.result_rows(res_cpu, 6)
    # filter kernel generic path too
    survivors = eng.filter(h, filt, want_row_ids=True)
    assert survivors == res_cpu.rows_passed
    eng.free(h)


def test_large_dict_value_fallback(eng):
    """Dict with >64 entries: filters can't use ref masks; generic VALUE
    evaluation must agree with the oracle."""
    rng = np.random.default_rng(3)
    rows = 40000
    vals = rng.choice(np.arange(0, 500, dtype=np.int64) * 7, rows)
    grp = rng.choice(np.frombuffer(b"AB", dtype=np.uint8), rows)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8), (abi.T_CHAR, 0, 0, 1)])
    blocks = []
    for i in range(0, rows, 2000):
        blocks.append(oracle.encode_block(
            schema, [vals[i:i+2000].view(np.uint8), grp[i:i+2000]],
            [abi.ENC_DICT, abi.ENC_DICT]))
    bs = _manual_blockset(schema, blocks)
    filt = abi.make_filter([dict(col=0, op=abi.OP_BT, lo=700, hi=2100)])
    agg = abi.make_agg([1], [dict(kind=abi.AGG_SUM, col_a=0),
                             dict(kind=abi.AGG_COUNT)])
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    h = eng.load(bs)
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    assert abi.result_rows(res_gpu, 2) == abi.result_rows(res_cpu, 2)
    eng.free(h)


def test_rle_agg_input(eng):
    """RLE-encoded column as an aggregate input (slow decode path in the
    aggregate passes)."""
    rng = np.random.default_rng(8)
    rows = 30000
    vals = np.repeat(rng.integers(0, 50, rows // 100 + 1, dtype=np.int64),
                     100)[:rows]
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    blocks = []
    for i in range(0, rows, 1500):
        blocks.append(oracle.encode_block(
            schema, [np.ascontiguousarray(vals[i:i+1500]).view(np.uint8)],
            [abi.ENC_RLE]))
    bs = _manual_blockset(schema, blocks)
    agg = abi.make_agg([], [dict(kind=abi.AGG_SUM, col_a=0),
                            dict(kind=abi.AGG_MIN, col_a=0),
                            dict(kind=abi.AGG_MAX, col_a=0)])
    res_cpu = oracle.scan_filter_agg(bs, None, agg)
    h = eng.load(bs)
    res_gpu = eng.scan_filter_agg(h, None, agg)
    assert abi.result_rows(res_gpu, 3) == abi.result_rows(res_cpu, 3)
    eng.free(h)


def test_filter_ops_on_date_intdiff(eng):
    """Signed-date domains through INTEGER_BASE_DIFF range lowering,
    including operands far outside the block's domain."""
    li = oracle.Lineitem(3, 60000, seed=21)
    h = eng.load(li.bs)
    lo = oracle.date_days(1995, 6, 17)
    for op, args in [(abi.OP_EQ, dict(lo=lo)), (abi.OP_NE, dict(lo=lo)),
                     (abi.OP_GE, dict(lo=lo)), (abi.OP_LT, dict(lo=-10**6)),
                     (abi.OP_GT, dict(lo=10**6)),
                     (abi.OP_BT, dict(lo=lo, hi=lo + 365))]:
        fd = abi.make_filter([dict(col=0, op=op, **args)])
        survivors = eng.filter(h, fd)
        res = oracle.scan_filter_agg(li.bs, fd, None)
        assert survivors == res.rows_passed, op
    eng.free(h)


def test_q1_parity_2m_rows(eng):
    """Larger-scale parity: full oracle comparison at 2M rows plus the
    size-independent invariants (sum of group counts == rows passed)."""
    li, h = _load_lineitem(eng, 4, 2_000_000)
    filt, agg = _q1_descs()
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(li.bs, filt, agg)
    assert abi.result_rows(res_gpu, 6) == abi.result_rows(res_cpu, 6)
    rows = abi.result_rows(res_gpu, 6)
    assert sum(rc for _, rc, _ in rows) == res_gpu.rows_passed
    assert all(cells[0] == rc for _, rc, cells in rows)  # COUNT(*) == rows
    # sum_disc_price <= sum_charge (tax >= 0), qty>0 -> sums positive
    for _, rc, cells in rows:
        assert 0 < cells[3] <= cells[4]
        assert cells[1] > 0 and cells[2] > 0
    eng.free(h)


def test_wide_rows_per_block(eng):
    """Blocks near the row_count cap (>> the 2048-row kernel window)."""
    rng = np.random.default_rng(17)
    rows = 60000
    vals = rng.integers(0, 1 << 40, rows, dtype=np.int64)
    grp = rng.choice(np.frombuffer(b"PQ", dtype=np.uint8), rows)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8), (abi.T_CHAR, 0, 0, 1)])
    block = oracle.encode_block(schema, [vals.view(np.uint8), grp],
                                [abi.ENC_RAW, abi.ENC_DICT])
    bs = _manual_blockset(schema, [block])
    filt = abi.make_filter([dict(col=0, op=abi.OP_GE, lo=1 << 39)])
    agg = abi.make_agg([1], [dict(kind=abi.AGG_SUM, col_a=0),
                             dict(kind=abi.AGG_MIN, col_a=0),
                             dict(kind=abi.AGG_MAX, col_a=0),
                             dict(kind=abi.AGG_COUNT)])
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    h = eng.load(bs)
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    assert abi.result_rows(res_gpu, 4) == abi.result_rows(res_cpu, 4)
    eng.free(h)


@pytest.mark.gpu
def test_jit_engages_on_q1_shape():
    """The hipRTC plan-specialized kernel must actually run for the
    flagship Q1 plan shape (guards against silent fallback to the generic
    kernels, which once hid a broken JIT compile)."""
    from oceanbase_amd.engine import GpuEngine
    li = oracle.Lineitem(4, 50_000, seed=11)
    eng = GpuEngine(0)
    h = eng.load(li.bs)
    filt = abi.make_filter(
        [dict(col=6, op=abi.OP_LE, lo=oracle.date_days(1998, 9, 2))])
    aggs = [dict(kind=abi.AGG_COUNT), dict(kind=abi.AGG_SUM, col_a=0),
            dict(kind=abi.AGG_SUM_PROD2, col_a=1, col_b=2),
            dict(kind=abi.AGG_SUM_PROD3, col_a=1, col_b=2, col_c=3)]
    agg = abi.make_agg([4, 5], aggs)
    res = eng.scan_filter_agg(h, filt, agg)
    assert eng.last_jit(), "JIT kernel did not engage for the Q1 shape"
    ores = oracle.scan_filter_agg(li.bs, filt, agg)
    assert res.rows_passed == ores.rows_passed
    assert sorted(abi.result_rows(res, 4)) == sorted(
        abi.result_rows(ores, 4))
    eng.close()


@pytest.mark.gpu
def test_jit_multi_window_blocks():
    """Blocks with more than 2048 rows split into LDS windows inside one
    kernel launch; exercise that path through the JIT kernel (narrow
    columns keep 4000-row blocks under the LDS stage size)."""
    from oceanbase_amd.engine import GpuEngine
    rows = 12000
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, 1), (abi.T_INT, 0, 0, 8)])
    rng = np.random.default_rng(77)
    flag = rng.choice(np.frombuffer(b"ANR", dtype=np.uint8), rows)
    qty = rng.integers(1, 50, rows, dtype=np.int64)
    blocks = []
    for s in range(0, rows, 4000):
        e = s + 4000
        blocks.append(oracle.encode_block(
            schema, [flag[s:e].copy(), qty[s:e].view(np.uint8)],
            [abi.ENC_DICT, abi.ENC_DICT], None))
    bs = _manual_blockset(schema, blocks)
    bs.total_rows = rows
    eng = GpuEngine(0)
    h = eng.load(bs)
    filt = abi.make_filter([dict(col=1, op=abi.OP_LT, lo=40)])
    agg = abi.make_agg([0], [dict(kind=abi.AGG_COUNT),
                             dict(kind=abi.AGG_SUM, col_a=1)])
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    assert eng.last_jit(), "expected the JIT path (narrow dict columns)"
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    assert sorted(abi.result_rows(res_gpu, 2)) == sorted(
        abi.result_rows(res_cpu, 2))
    eng.close()


@pytest.mark.gpu
def test_pipeline_agg_path_parity(eng):
    """OBX_PIPELINE_AGG=1 switches to the filter->group->agg kernel DAG;
    results must match the fused/JIT path bit-exactly."""
    import os as _os
    li = oracle.Lineitem(4, 60000, seed=9)
    h = eng.load(li.bs)
    filt, agg = _q1_descs()
    base = eng.scan_filter_agg(h, filt, agg)
    _os.environ["OBX_PIPELINE_AGG"] = "1"
    try:
        piped = eng.scan_filter_agg(h, filt, agg)
    finally:
        del _os.environ["OBX_PIPELINE_AGG"]
    assert base.rows_passed == piped.rows_passed
    assert sorted(abi.result_rows(base, 6)) == sorted(
        abi.result_rows(piped, 6))


@pytest.mark.gpu
def test_stored_minmax_skip_index_parity():
    """Stored min/max skip index (ObSSTableIndexFilter equivalent): RAW
    fixed columns cluster by block here, so selective range predicates
    prune most blocks via the bounds captured at load (k_col_minmax ->
    k_lower_leaves NONE verdicts); results must equal the oracle and the
    all-pass/none refinements must not change survivors."""
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(3)
    rows_pb, nblocks = 2000, 24
    blks, av = [], []
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8),
                                 (abi.T_INT, 0, 19, 8)])
    for b in range(nblocks):
        # values cluster per block: block b holds [b*1000, b*1000+999]
        a = rng.integers(b * 1000, b * 1000 + 1000, rows_pb).astype(np.int64)
        x = rng.integers(-10**6, 10**6, rows_pb).astype(np.int64)
        blks.append(oracle.encode_block(
            schema, [a.view(np.uint8), x.view(np.uint8)],
            [abi.ENC_RAW, abi.ENC_RAW]))
        av.append(a)
    bs = _manual_blockset(schema, blks)
    bs.total_rows = rows_pb * nblocks
    eng = GpuEngine(0)
    h = eng.load(bs)
    for lo, hi in ((5000, 6999), (0, 999), (23_000, 23_999), (40_000, 41_000),
                   (2500, 2503)):
        filt = abi.make_filter([dict(col=0, op=abi.OP_BT, lo=lo, hi=hi)])
        got = eng.filter(h, filt)
        exp = sum(int(((a >= lo) & (a <= hi)).sum()) for a in av)
        assert got == exp, (lo, hi)
    eng.close()


def test_q1_with_minmax_takes_v2_jit(eng):
    """MIN/MAX aggregates no longer force the generic fallback: CTX
    columns accumulate in LDS CAS cells, dict columns read their min/max
    off the histogram at flush — all bit-exact vs the oracle."""
    li, h = _load_lineitem(eng, 4, 150000)
    filt = abi.make_filter(
        [dict(col=6, op=abi.OP_LE, lo=oracle.date_days(1998, 9, 2))])
    aggs = [dict(kind=abi.AGG_COUNT),
            dict(kind=abi.AGG_SUM, col_a=1),
            dict(kind=abi.AGG_MIN, col_a=1),   # CTX raw8 -> wmm
            dict(kind=abi.AGG_MAX, col_a=1),
            dict(kind=abi.AGG_MIN, col_a=0),   # dict col -> hist flush
            dict(kind=abi.AGG_MAX, col_a=0)]
    agg = abi.make_agg([4, 5], aggs)
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    import ctypes as Ct
    assert eng._lib.obx_gpu_last_jit(eng._ctx) == 2  # persistent v2 ran
    res_cpu = oracle.scan_filter_agg(li.bs, filt, agg)
    assert abi.result_rows(res_gpu, 6) == abi.result_rows(res_cpu, 6)
    eng.free(h)
