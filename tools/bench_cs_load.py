"""CS GPU-native load-path throughput (the ObCSMicroBlockTransformer
equivalent): host metadata parse + k_cs_decode device stream decode into
the engine's scan layout, then a scan over the loaded handle.

Run on a GPU box:  python tools/bench_cs_load.py --rows 500000
Prints one JSON line: encoded bytes, load (parse+H2D+decode) time,
decode GB/s, and the post-load scan time.
"""
import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from oceanbase_amd import abi  # noqa: E402
from oceanbase_amd.engine import GpuEngine  # noqa: E402
from test_cs_block import _enc, _int_col, _str_col  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=500_000)
    ap.add_argument("--rows-per-block", type=int, default=3000)
    args = ap.parse_args()

    rng = np.random.default_rng(7)
    nblocks = (args.rows + args.rows_per_block - 1) // args.rows_per_block
    blocks = []
    t0 = time.time()
    for _ in range(nblocks):
        n = args.rows_per_block
        qty = rng.integers(1, 51, n).astype(np.int64)
        price = rng.integers(90000, 10**7, n).astype(np.int64)
        disc = rng.integers(0, 11, n).astype(np.int64)
        flag = [b"A" if x == 0 else b"N" if x == 1 else b"R"
                for x in rng.integers(0, 3, n)]
        date = rng.integers(8000, 10600, n).astype(np.int64)
        blocks.append(_enc(n, [
            _int_col(qty, dict_=True),
            _int_col(price, enc=6),       # SIMD_FIXEDPFOR
            _int_col(disc, dict_=True),
            _str_col(flag, dict_=True),
            _int_col(date, enc=5),        # DOUBLE_DELTA_ZIGZAG_PFOR
        ]))
    gen_s = time.time() - t0
    total_bytes = sum(len(b) for b in blocks)
    specs = [(abi.T_INT, 0, 19, 8), (abi.T_INT, 0, 19, 8),
             (abi.T_INT, 0, 19, 8), (abi.T_CHAR, 0, 0, 1),
             (abi.T_INT, 0, 19, 8)]
    eng = GpuEngine()
    schema = (abi.ColSchema * len(specs))()
    for i, (t, sc, p, ln) in enumerate(specs):
        schema[i].obj_type, schema[i].scale = t, sc
        schema[i].precision, schema[i].len = p, ln
    # warm load (allocs, module load), then timed loads
    h = eng.load_cs(blocks, schema)
    eng.free(h)
    times = []
    for i in range(3):
        t1 = time.time()
        h = eng.load_cs(blocks, schema)
        times.append(time.time() - t1)
        if i < 2:
            eng.free(h)
    load_s = min(times)
    filt = abi.make_filter([dict(col=4, op=abi.OP_LE, lo=10471)])
    agg = abi.make_agg([3], [dict(kind=abi.AGG_COUNT),
                             dict(kind=abi.AGG_SUM, col_a=0),
                             dict(kind=abi.AGG_SUM_PROD2, col_a=1,
                                  col_b=2)])
    eng.scan_filter_agg(h, filt, agg)  # warm
    t2 = time.time()
    res = eng.scan_filter_agg(h, filt, agg)
    scan_s = time.time() - t2
    print(json.dumps({
        "rows": args.rows, "cs_blocks": nblocks,
        "encoded_bytes": total_bytes,
        "gen_seconds": round(gen_s, 2),
        "load_seconds_min_of_3": round(load_s, 4),
        "load_GBps": round(total_bytes / load_s / 1e9, 2),
        "load_Mrows_per_s": round(args.rows / load_s / 1e6, 1),
        "scan_ms": round(scan_s * 1e3, 3),
        "rows_passed": res.rows_passed,
    }))
    eng.free(h)


if __name__ == "__main__":
    main()
