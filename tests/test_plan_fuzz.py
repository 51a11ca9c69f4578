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
        leaves = [dict(col=6, op=abi.OP_GE, lo=0)]
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
