"""Randomized AND/OR combine-program parity (CPU): the oracle's postfix
program evaluation (combine_leaves, mirroring
ObPushdownFilterExecutor::execute AND/OR semantics,
ob_pushdown_filter.cpp:1559) against an independent Python evaluation
over pymodel-decoded values."""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from oceanbase_amd import abi, oracle  # noqa: E402
import pymodel  # noqa: E402


def _random_program(rng, n_leaves):
    """Random valid postfix program (every leaf may repeat; ends with one
    value on the stack)."""
    stack = 1
    prog = [int(rng.integers(0, n_leaves))]
    steps = int(rng.integers(1, 10))
    for _ in range(steps):
        if stack >= 2 and rng.random() < 0.5:
            prog.append(int(rng.choice([abi.TOK_AND, abi.TOK_OR])))
            stack -= 1
        else:
            prog.append(int(rng.integers(0, n_leaves)))
            stack += 1
        if stack >= 8:
            break
    while stack > 1:
        prog.append(int(rng.choice([abi.TOK_AND, abi.TOK_OR])))
        stack -= 1
    return prog


@pytest.mark.parametrize("seed", range(25))
def test_program_parity_oracle_vs_pymodel(seed):
    rng = np.random.default_rng(1000 + seed)
    rows = int(rng.integers(64, 2500))
    n_leaves = int(rng.integers(2, 6))
    v = rng.integers(-100, 100, rows).astype(np.int64)
    nulls = None
    if rng.random() < 0.5:
        nulls = np.zeros((rows + 7) // 8, dtype=np.uint8)
        for r in rng.choice(rows, max(1, rows // 11), replace=False):
            nulls[r // 8] |= 1 << (r % 8)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    enc = int(rng.choice([abi.ENC_RAW, abi.ENC_DICT, abi.ENC_AUTO]))
    blk = oracle.encode_block(schema, [v.view(np.uint8)], [enc], [nulls])

    leaves = []
    for _ in range(n_leaves):
        op = int(rng.choice([abi.OP_EQ, abi.OP_LE, abi.OP_LT, abi.OP_GE,
                             abi.OP_GT, abi.OP_NE, abi.OP_BT, abi.OP_NU,
                             abi.OP_NN]))
        lo = int(rng.integers(-100, 100))
        hi = lo + int(rng.integers(0, 50))
        leaves.append(dict(col=0, op=op, lo=lo, hi=hi))
    prog = _random_program(rng, n_leaves)
    filt = abi.make_filter(leaves, prog=prog)
    bits, passed = oracle.filter_block(schema, 1, blk, filt)

    vals = pymodel.Block(blk, [(abi.T_INT, 0, 19, 8)]).decode_col(0)
    want = 0
    for r in range(rows):
        stack = []
        for t in prog:
            if t < n_leaves:
                lf = leaves[t]
                stack.append(bool(pymodel.eval_leaf(
                    lf["op"], vals[r], lf["lo"], lf["hi"], [],
                    pymodel.SC_INT, 8)))
            elif t == abi.TOK_AND:
                b2, a2 = stack.pop(), stack.pop()
                stack.append(a2 and b2)
            else:
                b2, a2 = stack.pop(), stack.pop()
                stack.append(a2 or b2)
        res = stack[0]
        want += bool(res)
        got = (bits[r // 8] >> (r % 8)) & 1
        assert bool(got) == bool(res), (seed, r)
    assert passed == want


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(5))
def test_program_parity_gpu(seed):
    """Same randomized plans through the GPU prog kernels vs the oracle."""
    from oceanbase_amd.engine import GpuEngine
    from test_gpu_parity import _manual_blockset
    rng = np.random.default_rng(1000 + seed)
    rows = int(rng.integers(64, 2500))
    n_leaves = int(rng.integers(2, 6))
    v = rng.integers(-100, 100, rows).astype(np.int64)
    nulls = None
    if rng.random() < 0.5:
        nulls = np.zeros((rows + 7) // 8, dtype=np.uint8)
        for r in rng.choice(rows, max(1, rows // 11), replace=False):
            nulls[r // 8] |= 1 << (r % 8)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    enc = int(rng.choice([abi.ENC_RAW, abi.ENC_DICT, abi.ENC_AUTO]))
    blk = oracle.encode_block(schema, [v.view(np.uint8)], [enc], [nulls])
    leaves = []
    for _ in range(n_leaves):
        op = int(rng.choice([abi.OP_EQ, abi.OP_LE, abi.OP_LT, abi.OP_GE,
                             abi.OP_GT, abi.OP_NE, abi.OP_BT, abi.OP_NU,
                             abi.OP_NN]))
        lo = int(rng.integers(-100, 100))
        hi = lo + int(rng.integers(0, 50))
        leaves.append(dict(col=0, op=op, lo=lo, hi=hi))
    prog = _random_program(rng, n_leaves)
    filt = abi.make_filter(leaves, prog=prog)
    _, passed = oracle.filter_block(schema, 1, blk, filt)
    bs = _manual_blockset(schema, [blk])
    bs.total_rows = rows
    eng = GpuEngine(0)
    h = eng.load(bs)
    assert eng.filter(h, filt) == passed
    agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT),
                            dict(kind=abi.AGG_SUM, col_a=0)])
    res_gpu = eng.scan_filter_agg(h, filt, agg)
    res_cpu = oracle.scan_filter_agg(bs, filt, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed == passed
    assert sorted(abi.result_rows(res_gpu, 2)) == sorted(
        abi.result_rows(res_cpu, 2))
    eng.close()
