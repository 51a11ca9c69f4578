"""AND/OR filter-tree combine parity (ObPushdownFilterExecutor::execute,
ob_pushdown_filter.cpp:1559-1632): postfix combine programs over white
leaves, CPU oracle vs naive Python eval, and GPU vs oracle."""
import numpy as np
import pytest

from oceanbase_amd import abi, oracle
import pymodel

RNG = np.random.default_rng(77)


def _mk(vals, dtype):
    return np.ascontiguousarray(np.asarray(vals, dtype=dtype)).view(np.uint8)


def _naive(vals_per_leaf, leaves, prog, schema_tuples):
    rows = len(vals_per_leaf[0])
    out = []
    for r in range(rows):
        res = []
        for i, lf in enumerate(leaves):
            c = lf["col"]
            sc = pymodel.store_class(schema_tuples[c][0])
            res.append(pymodel.eval_leaf(lf["op"], vals_per_leaf[i][r],
                                         lf.get("lo", 0), lf.get("hi", 0),
                                         lf.get("in_list", []), sc,
                                         schema_tuples[c][3]))
        stack = []
        for t in prog:
            if t < len(leaves):
                stack.append(res[t])
            elif t == abi.TOK_AND:
                b = stack.pop(); stack[-1] = stack[-1] and b
            else:
                b = stack.pop(); stack[-1] = stack[-1] or b
        out.append(stack[0])
    return out


@pytest.mark.parametrize("prog_shape", ["or2", "or_and", "and_or_or"])
def test_oracle_or_trees(prog_shape):
    rows = 4000
    a = RNG.integers(0, 100, rows, dtype=np.int64)
    b = RNG.integers(0, 100, rows, dtype=np.int64)
    c = RNG.integers(0, 100, rows, dtype=np.int64)
    schema_t = [(abi.T_INT, 0, 19, 8)] * 3
    schema = oracle.make_schema(schema_t)
    blk = oracle.encode_block(schema, [_mk(a, np.int64), _mk(b, np.int64),
                                       _mk(c, np.int64)],
                              [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_INT_DIFF])
    leaves = [dict(col=0, op=abi.OP_LT, lo=30),
              dict(col=1, op=abi.OP_GE, lo=70),
              dict(col=2, op=abi.OP_BT, lo=40, hi=60)]
    progs = {"or2": [0, 1, abi.TOK_OR],
             "or_and": [0, 1, abi.TOK_OR, 2, abi.TOK_AND],
             "and_or_or": [0, 2, abi.TOK_AND, 1, abi.TOK_OR]}
    prog = progs[prog_shape]
    fd = abi.make_filter(leaves, prog=prog)
    bits, pc = oracle.filter_block(schema, 3, blk, fd)
    pb = pymodel.Block(blk, schema_t)
    vals = [pb.decode_col(lf["col"]) for lf in leaves]
    expect = _naive(vals, leaves, prog, schema_t)
    assert pc == sum(expect)
    for r, e in enumerate(expect):
        assert bool((bits[r >> 3] >> (r & 7)) & 1) == bool(e), r


def test_oracle_rejects_bad_program():
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    blk = oracle.encode_block(schema,
                              [_mk(RNG.integers(0, 9, 100, dtype=np.int64),
                                   np.int64)], [abi.ENC_RAW])
    fd = abi.make_filter([dict(col=0, op=abi.OP_LT, lo=5)],
                         prog=[0, 0])  # leaves left on stack
    with pytest.raises(RuntimeError):
        oracle.filter_block(schema, 1, blk, fd)


@pytest.mark.gpu
@pytest.mark.parametrize("case_seed", range(10))
def test_gpu_or_tree_parity(case_seed):
    """Random OR/AND trees over a lineitem config-3 set: GPU vs oracle."""
    from oceanbase_amd.engine import GpuEngine
    rng = np.random.default_rng(500 + case_seed)
    li = oracle.Lineitem(3, 30000, seed=7)
    cutoff = oracle.date_days(1996, 1, 1)
    leaves = [dict(col=0, op=abi.OP_LE, lo=cutoff),
              dict(col=1, op=abi.OP_LT, lo=int(rng.integers(5, 45))),
              dict(col=2, op=abi.OP_EQ, lo=ord("F")),
              dict(col=3, op=abi.OP_BT, lo=2, hi=int(rng.integers(3, 10)))]
    # random postfix tree over the 4 leaves
    prog = [0, 1]
    prog.append(int(rng.choice([abi.TOK_AND, abi.TOK_OR])))
    prog += [2]
    prog.append(int(rng.choice([abi.TOK_AND, abi.TOK_OR])))
    prog += [3]
    prog.append(int(rng.choice([abi.TOK_AND, abi.TOK_OR])))
    fd = abi.make_filter(leaves, prog=prog)
    res_cpu = oracle.scan_filter_agg(li.bs, fd, None)
    eng = GpuEngine(0)
    h = eng.load(li.bs)
    survivors = eng.filter(h, fd)
    assert survivors == res_cpu.rows_passed, (case_seed, prog)
    # and through the fused agg path (scalar count)
    agg = abi.make_agg([], [dict(kind=abi.AGG_COUNT)])
    res_gpu = eng.scan_filter_agg(h, fd, agg)
    assert res_gpu.rows_passed == res_cpu.rows_passed
    rows = abi.result_rows(res_gpu, 1)
    if res_cpu.rows_passed:
        assert rows[0][2][0] == res_cpu.rows_passed
    eng.close()


@pytest.mark.gpu
def test_gpu_or_bitmap_takes_filter_jit():
    """A staged-eligible OR program (two narrow leaf streams) compiles
    into the filter JIT (codegen-folded expression) and matches the
    oracle bitmap bit-for-bit."""
    from oceanbase_amd.engine import GpuEngine
    li = oracle.Lineitem(3, 60000, seed=9)
    cutoff = oracle.date_days(1996, 6, 1)
    fd = abi.make_filter([dict(col=0, op=abi.OP_LE, lo=cutoff),
                          dict(col=1, op=abi.OP_LT, lo=8)],
                         prog=[0, 1, abi.TOK_OR])
    res_cpu = oracle.scan_filter_agg(li.bs, fd, None)
    eng = GpuEngine(0)
    h = eng.load(li.bs)
    survivors = eng.filter(h, fd)
    assert survivors == res_cpu.rows_passed
    assert eng._lib.obx_gpu_last_jit(eng._ctx) == 2
    eng.close()
