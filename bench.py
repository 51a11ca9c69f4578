#!/usr/bin/env python3
"""bench.py — measures the BASELINE.json metric: rows/sec (+ achieved HBM
GB/s) of the TPC-H Q1 lineitem scan->filter->aggregate hot path on MI355X.

Contract: `python bench.py --gpus N --steps K --warmup W`. One rank per GPU
(torchrun for N>1; reads RANK/LOCAL_RANK/WORLD_SIZE). A "step" is one full
pass of the hot path over the synthetic workload: fused GPU
scan->filter->group-by over the rank's microblock shard + the partial-
aggregate exchange across ranks (the reference's 2-phase group-by exchange;
SURVEY.md §2 collective inventory) + local merge. W untimed warmups, exactly
K timed steps between barrier+synchronize pairs, MAX over ranks, rank 0
prints ONE JSON line.

Workloads (BASELINE.json configs): q1 (default, config 4 = TPC-H Q1 SF=100),
q6 (config 5 shape), filter-int64 (config 2), decode-filter (config 3).
Weak scaling: each rank owns a full per-GPU shard of --rows rows (row-range
sharded generation is byte-deterministic).

Inputs are resident in HBM before the timed region starts (staged once at
load). The oracle appears ONLY in the cpu_baseline leg (rank 0, N=1).
"""
import argparse
import ctypes as C
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np  # noqa: E402

from oceanbase_amd import abi, oracle  # noqa: E402

HBM_PEAK_BYTES = 8.0e12  # MI355X spec peak (measured ceiling ~6.3 TB/s;
                         # /opt/skills/guides/MI355X_MICROARCH.md)

WORKLOADS = {
    "q1": dict(config=4, default_rows=600_000_000,
               name="tpch_q1_sf100_lineitem_scan_filter_agg"),
    "q6": dict(config=6, default_rows=600_000_000,
               name="tpch_q6_lineitem_filter_sum"),
    "filter-int64": dict(config=2, default_rows=100_000_000,
                         name="raw_int64_filter_100m"),
    "decode-filter": dict(config=3, default_rows=60_000_000,
                          name="sf10_4col_decode_filter"),
}


def build_descs(workload):
    if workload == "q1":
        filt = abi.make_filter(
            [dict(col=6, op=abi.OP_LE, lo=oracle.date_days(1998, 9, 2))])
        aggs = [
            dict(kind=abi.AGG_COUNT),
            dict(kind=abi.AGG_SUM, col_a=0),
            dict(kind=abi.AGG_SUM, col_a=1),
            dict(kind=abi.AGG_SUM_PROD2, col_a=1, col_b=2),
            dict(kind=abi.AGG_SUM_PROD3, col_a=1, col_b=2, col_c=3),
            dict(kind=abi.AGG_SUM, col_a=2),
        ]
        n = int(os.environ.get("OBX_Q1_AGGS", "6"))  # diagnostics knob
        aggs = aggs[:n]
        agg = abi.make_agg([4, 5], aggs)
        return filt, agg, len(aggs)
    if workload == "q6":
        d94, d95 = oracle.date_days(1994, 1, 1), oracle.date_days(1995, 1, 1)
        filt = abi.make_filter([
            dict(col=0, op=abi.OP_GE, lo=d94),
            dict(col=0, op=abi.OP_LT, lo=d95),
            dict(col=1, op=abi.OP_BT, lo=5, hi=7),
            dict(col=2, op=abi.OP_LT, lo=2400)])
        agg = abi.make_agg([], [dict(kind=abi.AGG_SUM_MUL, col_a=3, col_b=1)])
        return filt, agg, 1
    if workload == "filter-int64":
        return abi.make_filter([dict(col=0, op=abi.OP_LT, lo=24)]), None, 0
    if workload == "decode-filter":
        filt = abi.make_filter(
            [dict(col=0, op=abi.OP_LE, lo=oracle.date_days(1998, 9, 2))])
        return filt, None, 0
    raise ValueError(workload)


def result_to_bytes(res):
    return bytes(res)  # ctypes struct -> raw bytes


_SUM_FAMILY = {abi.AGG_COUNT, abi.AGG_SUM, abi.AGG_SUM_PROD2,
               abi.AGG_SUM_PROD3, abi.AGG_SUM_MUL}


def merge_results(blobs, n_aggs, kinds=None):
    """Merge serialized AggResults (Python ints: exact 256-bit adds).

    Cells are added for sum-family aggregates and min/max-folded for
    MIN/MAX; `kinds` (list of abi.AGG_* per aggregate) must be passed
    when any non-sum aggregate is present — without it, a MIN/MAX cell
    reaching this merge fails loudly instead of being summed wrong."""
    if kinds is not None:
        assert len(kinds) == n_aggs
    groups = {}
    scanned = passed = 0
    for blob in blobs:
        r = abi.AggResult.from_buffer_copy(blob)
        scanned += r.rows_scanned
        passed += r.rows_passed
        for key, cnt, cells in abi.result_rows(r, n_aggs):
            g = groups.setdefault(key, [0] + [None] * n_aggs)
            g[0] += cnt
            for a in range(n_aggs):
                kind = kinds[a] if kinds is not None else abi.AGG_SUM
                if kind in _SUM_FAMILY:
                    g[a + 1] = (g[a + 1] or 0) + cells[a]
                elif kind == abi.AGG_MIN:
                    g[a + 1] = cells[a] if g[a + 1] is None \
                        else min(g[a + 1], cells[a])
                elif kind == abi.AGG_MAX:
                    g[a + 1] = cells[a] if g[a + 1] is None \
                        else max(g[a + 1], cells[a])
                else:
                    raise AssertionError(
                        f"aggregate kind {kind} has no distributed merge")
    return dict(sorted(groups.items())), scanned, passed


def measure_traffic(args):
    """HBM traffic of the dominant kernel, per launch, from TCC fabric
    counters: re-runs this bench (3 launches) under `rocprofv3 --pmc`
    in a subprocess and parses the per-dispatch counter CSV. Read bytes
    = RDREQ x 64 B x 2 (the gfx950 half-count of wide coalesced reads,
    MI355X_MICROARCH.md §HBM, calibrated on filter-int64 in round 1);
    write bytes = WRREQ x 64 B (uncalibrated). Returns a dict or None."""
    import csv
    import glob
    import shutil
    import subprocess
    import tempfile
    rocprof = shutil.which("rocprofv3")
    if rocprof is None:
        return None
    me = os.path.abspath(__file__)
    with tempfile.TemporaryDirectory(prefix="obx_pmc_") as td:
        cmd = [rocprof, "--pmc", "TCC_EA0_RDREQ", "TCC_EA0_WRREQ",
               "--output-format", "csv", "-d", td, "--", sys.executable, me,
               "--workload", args.workload, "--rows", str(args.rows or 0),
               "--block-bytes", str(args.block_bytes),
               "--seed", str(args.seed), "--steps", "2", "--warmup", "1",
               "--no-cpu-baseline", "--no-traffic"]
        env = dict(os.environ, TMPDIR=td)
        try:
            subprocess.run(cmd, cwd=td, env=env, check=True,
                           stdout=subprocess.DEVNULL,
                           stderr=subprocess.DEVNULL, timeout=600)
        except Exception:
            return None
        per_kernel = {}  # name -> {counter: total, "n": dispatches}
        for f in glob.glob(os.path.join(td, "**", "*counter_collection.csv"),
                           recursive=True):
            with open(f, newline="") as fh:
                for row in csv.DictReader(fh):
                    name = row.get("Kernel_Name", "")
                    cname = row.get("Counter_Name", "")
                    try:
                        val = float(row.get("Counter_Value", "0"))
                    except ValueError:
                        continue
                    k = per_kernel.setdefault(name, {})
                    k[cname] = k.get(cname, 0.0) + val
                    if cname == "TCC_EA0_RDREQ":
                        k["n"] = k.get("n", 0) + 1
        main_k = None
        for name, k in per_kernel.items():
            if not name.startswith("k_"):
                continue
            if main_k is None or k.get("TCC_EA0_RDREQ", 0) > \
                    per_kernel[main_k].get("TCC_EA0_RDREQ", 0):
                main_k = name
        if main_k is None or per_kernel[main_k].get("n", 0) == 0:
            return None
        k = per_kernel[main_k]
        n = k["n"]
        return dict(kernel=main_k.split("(")[0],
                    read_bytes=k.get("TCC_EA0_RDREQ", 0) * 64 * 2 / n,
                    write_bytes=k.get("TCC_EA0_WRREQ", 0) * 64 / n,
                    launches=n)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--workload", default="q1", choices=sorted(WORKLOADS))
    ap.add_argument("--rows", type=int, default=0,
                    help="rows per GPU (default: workload's full size)")
    ap.add_argument("--block-bytes", type=int, default=16384)
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--no-traffic", action="store_true",
                    help="skip the rocprofv3 TCC-traffic leg")
    ap.add_argument("--trace", action="store_true",
                    help="per-step device-time table (prep/main kernel) — "
                         "the per-operator monitor rebuild (SURVEY §5)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(world, 1)

    wl = WORKLOADS[args.workload]
    rows_per_gpu = args.rows or wl["default_rows"]

    import torch
    dist = None
    # RCCL (backend "nccl") for real multi-GPU; OBX_DIST_BACKEND=gloo lets a
    # 1-GPU box smoke the full multi-rank path (exchange on CPU)
    backend = os.environ.get("OBX_DIST_BACKEND", "nccl")
    n_dev = torch.cuda.device_count() if torch.cuda.is_available() else 1
    device = local_rank % max(n_dev, 1)
    if world > 1:
        import torch.distributed as tdist
        tdist.init_process_group(backend=backend)
        if backend == "nccl":
            torch.cuda.set_device(device)
        dist = tdist

    # --- generate this rank's shard (deterministic by global row id) -------
    t0 = time.time()
    if world > 1:
        os.environ.setdefault("OBX_GEN_THREADS",
                              str(max(1, (os.cpu_count() or 8) // world)))
    # shard boundaries at block granularity: row_base multiple of rows/blk
    li = oracle.Lineitem(wl["config"], rows_per_gpu, seed=args.seed,
                         block_bytes=args.block_bytes,
                         row_base=rank * rows_per_gpu)
    gen_s = time.time() - t0

    from oceanbase_amd.engine import GpuEngine
    eng = GpuEngine(device)
    t0 = time.time()
    h = eng.load(li.bs)
    load_s = time.time() - t0

    filt, agg, n_aggs = build_descs(args.workload)

    def one_step():
        if agg is not None:
            res = eng.scan_filter_agg(h, filt, agg)
            if dist is not None:
                blob = result_to_bytes(res)
                t = torch.frombuffer(bytearray(blob), dtype=torch.uint8)
                if backend == "nccl":
                    t = t.cuda(device)
                outs = [torch.empty_like(t) for _ in range(world)]
                dist.all_gather(outs, t)
                kinds = [agg.aggs[i].kind for i in range(n_aggs)]
                merged = merge_results(
                    [bytes(o.cpu().numpy().tobytes()) for o in outs], n_aggs,
                    kinds)
                return merged
            return res
        else:
            eng.filter(h, filt, want_row_ids=False)
            return None

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        torch.cuda.synchronize() if torch.cuda.is_available() else None

    kernel_ms = []
    prep_ms = []
    for _ in range(args.warmup):
        one_step()
    jit = eng.last_jit()
    barrier_sync()
    t_start = time.time()
    for _ in range(args.steps):
        t_step = time.time()
        one_step()
        kernel_ms.append(eng.last_kernel_ms())
        prep_ms.append(eng.last_prep_ms())
        if args.trace and rank == 0:
            print(f"# step {len(kernel_ms)}: prep={prep_ms[-1]:.3f} ms "
                  f"main_kernel={kernel_ms[-1]:.3f} ms "
                  f"step_wall={(time.time()-t_step)*1e3:.3f} ms",
                  file=sys.stderr)
    barrier_sync()
    elapsed = time.time() - t_start

    # MAX over ranks
    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if backend == "nccl":
            t = t.cuda(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu().item())

    total_rows = rows_per_gpu * n_gpus
    ms_per_step = elapsed / args.steps * 1000.0
    rows_per_sec = total_rows / (elapsed / args.steps)

    # --- roofline (dominant kernel, algorithmic bytes / avg launch time) ---
    n_cols = li.n_cols
    header_bytes = li.n_blocks * (64 + 16 * n_cols)
    algo_bytes = li.total_bytes - header_bytes  # encoded column bytes/launch
    if args.workload == "filter-int64":
        algo_bytes += li.total_rows // 8  # result bitmap write
    avg_kms = sum(kernel_ms) / max(len(kernel_ms), 1)
    achieved = algo_bytes / (avg_kms / 1000.0) if avg_kms > 0 else 0.0
    roofline = dict(bound="hbm", achieved=achieved, peak=HBM_PEAK_BYTES,
                    unit="GB/s", frac=achieved / HBM_PEAK_BYTES,
                    traffic=None)
    # express achieved/peak in GB/s
    roofline["achieved"] = achieved / 1e9
    roofline["peak"] = HBM_PEAK_BYTES / 1e9

    # --- measured HBM traffic (TCC fabric counters, separate rocprofv3
    #     child run of the same workload; per launch of the main kernel) ----
    if rank == 0 and world <= 1 and not args.no_traffic \
            and os.environ.get("OBX_BENCH_TRAFFIC", "1") != "0":
        tr = measure_traffic(args)
        if tr is not None:
            roofline["traffic"] = tr["read_bytes"] + tr["write_bytes"]
            roofline["traffic_detail"] = dict(
                kernel=tr["kernel"],
                read_bytes=tr["read_bytes"], write_bytes=tr["write_bytes"],
                note="RDREQ*64B*2 (gfx950 wide-read half-count correction, "
                     "calibrated r01) + WRREQ*64B, per launch")

    # --- CPU baseline (oracle on host cores; fixed sample, 3 reps; rank0
    #     N=1). The oracle is -O3 -march=native auto-vectorized C, not the
    #     reference's hand-written AVX512 — stated in `sample`. ------------
    cpu_baseline = None
    if rank == 0 and world <= 1 and not args.no_cpu_baseline:
        cores = os.cpu_count() or 1
        rpb = max(li.total_rows // li.n_blocks, 1)
        nblk = min(max(int(150_000_000 // rpb), 1), li.n_blocks)
        sub = li.subset(nblk)
        rates = []
        srows = 0
        for _ in range(3):
            t0 = time.time()
            r = oracle.scan_filter_agg(sub, filt, agg, nthreads=0)
            dt = time.time() - t0
            srows = r.rows_scanned
            rates.append(srows / dt)
        rates.sort()
        spread = (rates[-1] - rates[0]) / rates[1] if rates[1] else 0.0
        cpu_baseline = dict(
            value=rates[1], unit="rows/s", cores=cores,
            kind="port",
            sample=f"fixed {srows}-row sample, median of 3 reps "
                   f"(spread {spread:.1%}); oracle = auto-vectorized C "
                   f"(-O3 -march=native), threaded")

    if rank == 0:
        line = dict(
            metric="rows/sec, TPC-H lineitem scan->filter->agg (microblock "
                   "decode + pushdown filter + group-by)",
            value=rows_per_sec,
            unit="rows/s",
            n_gpus=n_gpus,
            steps=args.steps,
            warmup=args.warmup,
            ms_per_step=ms_per_step,
            higher_is_better=True,
            scaling="weak",
            vs_baseline=None,  # no published per-path number
            dtype="int64",
            data="synthetic",
            jit=jit,
            config=dict(workload=wl["name"],
                        rows_per_gpu=rows_per_gpu,
                        total_rows=total_rows,
                        encoded_bytes_per_gpu=li.total_bytes,
                        n_blocks=li.n_blocks,
                        block_bytes=args.block_bytes,
                        gen_seconds=round(gen_s, 2),
                        h2d_seconds=round(load_s, 2),
                        kernel_ms_avg=round(avg_kms, 4),
                        prep_ms_avg=round(sum(prep_ms) / max(len(prep_ms), 1),
                                          4)),
            roofline=roofline,
            cpu_baseline=cpu_baseline,
        )
        print(json.dumps(line))
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
