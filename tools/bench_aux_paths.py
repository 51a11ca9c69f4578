"""Auxiliary-path timings: selection vectors (get_row_ids shape) and
whole-column projection decode (get_rows shape). Not contract metrics;
recorded in the §8 ledger."""
import json
import sys
import time

sys.path.insert(0, ".")
from oceanbase_amd import abi, oracle  # noqa: E402
from oceanbase_amd.engine import GpuEngine  # noqa: E402


def main():
    rows = 60_000_000
    li = oracle.Lineitem(3, rows, seed=42)
    eng = GpuEngine()
    h = eng.load(li.bs)
    filt = abi.make_filter(
        [dict(col=0, op=abi.OP_LE, lo=oracle.date_days(1998, 9, 2))])
    # selection vectors (row_ids + per-block counts)
    eng.filter(h, filt, want_row_ids=True)
    t, n = [], 5
    for _ in range(n):
        t0 = time.time()
        eng.filter(h, filt, want_row_ids=True)
        t.append(time.time() - t0)
    rid_ms = min(t) * 1e3
    rid_k = eng._lib.obx_gpu_last_kernel_ms(eng._ctx)
    # whole-block projection decode (all 4 cols, device-resident)
    import ctypes as C
    lib = eng._lib
    proj = (C.c_uint16 * 4)(0, 1, 2, 3)
    lib.obx_gpu_decode(eng._ctx, h, proj, 4)
    t0 = time.time()
    lib.obx_gpu_decode(eng._ctx, h, proj, 4)
    dec_ms = (time.time() - t0) * 1e3
    dec_k = eng._lib.obx_gpu_last_kernel_ms(eng._ctx)
    print(json.dumps({
        "rows": rows,
        "row_ids_wall_ms_min": round(rid_ms, 3),
        "row_ids_kernel_ms": round(rid_k, 3),
        "decode_4col_wall_ms": round(dec_ms, 3),
        "decode_4col_kernel_ms": round(dec_k, 3),
    }))
    eng.free(h)


if __name__ == "__main__":
    main()
