"""Randomized (seeded, deterministic) plan fuzz: many filter+aggregate
plans over lineitem-shaped data, each checked bit-exact GPU-vs-oracle,
with the JIT path and the generic path cross-checked against each other
(OBX_JIT=0 pins the precompiled generic kernels)."""
import os
import random

import pytest

from oceanbase_amd import abi, oracle

N_PLANS = 24


def _rand_plan(rng, n_cols=7):
    # columns: 0 qty(dict) 1 price(raw8) 2 disc(dict) 3 tax(dict)
    #          4 rf(dict) 5 ls(dict) 6 shipdate(intdiff)
    leaves = []
    for _ in range(rng.randint(0, 3)):
        col = rng.randrange(0, n_cols)
        op = rng.choice([abi.OP_EQ, abi.OP_LE, abi.OP_LT, abi.OP_GE,
                         abi.OP_GT, abi.OP_NE, abi.OP_BT])
        lo = rng.randint(-10, 11000)
        hi = lo + rng.randint(0, 3000)
        d = dict(col=col, op=op, lo=lo)
        if op == abi.OP_BT:
            d["hi"] = hi
        leaves.append(d)
    if not leaves:
        leaves = [dict(col=n_cols - 1, op=abi.OP_GE, lo=-10**9)]
    filt = abi.make_filter(leaves)
    kinds = [abi.AGG_COUNT, abi.AGG_SUM, abi.AGG_MIN, abi.AGG_MAX,
             abi.AGG_SUM_PROD2, abi.AGG_SUM_PROD3]
    aggs = [dict(kind=abi.AGG_COUNT)]
    for _ in range(rng.randint(1, 5)):
        k = rng.choice(kinds)
        a = dict(kind=k, col_a=rng.choice([0, 1, 2, 3]))
        if k in (abi.AGG_SUM_PROD2, abi.AGG_SUM_PROD3):
            a["col_b"] = rng.choice([0, 2, 3])
            if k == abi.AGG_SUM_PROD3:
                a["col_c"] = rng.choice([0, 2, 3])
        aggs.append(a)
    groups = rng.choice([[4], [5], [4, 5], [5, 4]])
    return filt, abi.make_agg(groups, aggs), len(aggs)


@pytest.mark.gpu
def test_plan_fuzz_gpu_vs_oracle_vs_generic():
    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine()
    li = oracle.Lineitem(4, 120000, seed=99)
    h = eng.load(li.bs)
    rng = random.Random(20260915)
    for i in range(N_PLANS):
        filt, agg, n_aggs = _rand_plan(rng)
        res_cpu = oracle.scan_filter_agg(li.bs, filt, agg)
        exp = abi.result_rows(res_cpu, n_aggs)
        res_jit = eng.scan_filter_agg(h, filt, agg)
        assert abi.result_rows(res_jit, n_aggs) == exp, f"plan {i} (jit)"
        os.environ["OBX_JIT"] = "0"
        try:
            res_gen = eng.scan_filter_agg(h, filt, agg)
        finally:
            del os.environ["OBX_JIT"]
        assert abi.result_rows(res_gen, n_aggs) == exp, f"plan {i} (generic)"
    eng.free(h)


def _null_rich_bs():
    import ctypes as C
    import numpy as np
    from test_group_capacity import _blockset
    rng = np.random.default_rng(59)
    rows_pb, nblocks = 1500, 5
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)] * 4 +
                                [(abi.T_CHAR, 0, 0, 1)] * 2)
    blocks = []
    for _ in range(nblocks):
        c0 = rng.integers(0, 40, rows_pb).astype(np.int64)      # dict
        c1 = rng.integers(-10**6, 10**6, rows_pb).astype(np.int64)  # raw
        c2 = rng.integers(0, 9, rows_pb).astype(np.int64)       # dict
        c3 = np.cumsum(rng.integers(0, 3, rows_pb)).astype(np.int64)  # diff
        g0 = (65 + rng.integers(0, 3, rows_pb)).astype(np.uint8)
        g1 = (97 + rng.integers(0, 2, rows_pb)).astype(np.uint8)
        nb_ = (rows_pb + 7) // 8
        nmaps = []
        for step in (7, 11, 13, 0, 0, 0):
            if step:
                m = np.zeros(nb_, dtype=np.uint8)
                for r in range(0, rows_pb, step):
                    m[r >> 3] |= 1 << (r & 7)
                nmaps.append(m)
            else:
                nmaps.append(None)
        blocks.append(oracle.encode_block(
            schema,
            [c0.view(np.uint8), c1.view(np.uint8), c2.view(np.uint8),
             c3.view(np.uint8), g0, g1],
            [abi.ENC_DICT, abi.ENC_RAW, abi.ENC_DICT, abi.ENC_INT_DIFF,
             abi.ENC_DICT, abi.ENC_DICT],
            nmaps))
    return _blockset(schema, blocks, rows_pb * nblocks)


@pytest.mark.gpu
def test_plan_fuzz_null_rich():
    """24 random plans over a null-rich mixed-encoding table (dict
    nulls, ext nulls, diff-coded), groups on two char dicts — GPU JIT
    and generic paths vs the oracle."""
    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine()
    bs = _null_rich_bs()
    h = eng.load(bs)
    rng = random.Random(31337)
    for i in range(N_PLANS):
        filt, agg, n_aggs = _rand_plan(rng, n_cols=4)
        # remap groups onto cols 4/5
        groups = rng.choice([[4], [5], [4, 5], [5, 4]])
        agg2 = abi.make_agg(groups,
                            [dict(kind=abi.AGG_COUNT)] +
                            [dict(kind=rng.choice(
                                [abi.AGG_COUNT, abi.AGG_SUM, abi.AGG_MIN,
                                 abi.AGG_MAX, abi.AGG_SUM_PROD2]),
                                  col_a=rng.choice([0, 1, 2, 3]),
                                  col_b=rng.choice([0, 2, 3]))
                             for _ in range(rng.randint(1, 4))])
        n_aggs = agg2.n_aggs
        res_cpu = oracle.scan_filter_agg(bs, filt, agg2)
        exp = abi.result_rows(res_cpu, n_aggs)
        res_jit = eng.scan_filter_agg(h, filt, agg2)
        assert abi.result_rows(res_jit, n_aggs) == exp, f"plan {i} (jit)"
        os.environ["OBX_JIT"] = "0"
        try:
            res_gen = eng.scan_filter_agg(h, filt, agg2)
        finally:
            del os.environ["OBX_JIT"]
        assert abi.result_rows(res_gen, n_aggs) == exp, \
            f"plan {i} (generic)"
    eng.free(h)


def test_out_of_range_columns_rejected():
    """Out-of-range filter/group/aggregate columns return
    OBX_INVALID_ARGUMENT (the oracle used to walk off the schema —
    found by this fuzz's own out-of-range fallback leaf)."""
    bs = _null_rich_bs()
    ok_agg = abi.make_agg([4], [dict(kind=abi.AGG_COUNT)])
    bad_filt = abi.make_filter([dict(col=6, op=abi.OP_GE, lo=0)])
    with pytest.raises(RuntimeError):
        oracle.scan_filter_agg(bs, bad_filt, ok_agg)
    with pytest.raises(RuntimeError):
        oracle.scan_filter_agg(bs, None, abi.make_agg(
            [9], [dict(kind=abi.AGG_COUNT)]))
    with pytest.raises(RuntimeError):
        oracle.scan_filter_agg(bs, None, abi.make_agg(
            [4], [dict(kind=abi.AGG_SUM, col_a=77)]))
    with pytest.raises(RuntimeError):
        oracle.scan_filter_agg(bs, None, abi.make_agg(
            [4], [dict(kind=abi.AGG_SUM_PROD2, col_a=0, col_b=66)]))


@pytest.mark.gpu
def test_out_of_range_columns_rejected_gpu():
    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine()
    bs = _null_rich_bs()
    h = eng.load(bs)
    with pytest.raises(RuntimeError):
        eng.filter(h, abi.make_filter([dict(col=6, op=abi.OP_GE, lo=0)]))
    with pytest.raises(RuntimeError):
        eng.scan_filter_agg(h, None, abi.make_agg(
            [9], [dict(kind=abi.AGG_COUNT)]))
    eng.free(h)


def test_bad_scale_rejected():
    """Out-of-range decimal scale in the ABI schema indexes the P10
    tables: both sides must reject it."""
    import numpy as np
    from test_group_capacity import _blockset
    rng = np.random.default_rng(79)
    schema = oracle.make_schema([(abi.T_DECIMAL_INT, 2, 15, 8)])
    v = rng.integers(0, 100, 500).astype(np.int64)
    blk = oracle.encode_block(schema, [v.view(np.uint8)], [abi.ENC_RAW])
    bs = _blockset(schema, [blk], 500)
    schema[0].scale = 77  # hostile schema byte
    agg = abi.make_agg([], [dict(kind=abi.AGG_SUM, col_a=0)])
    with pytest.raises(RuntimeError):
        oracle.scan_filter_agg(bs, None, agg)
