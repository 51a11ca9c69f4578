"""Multi-rank partial-aggregate exchange logic on CPU (gloo, world_size 2).

Covers the distributed path the 8-GPU bench uses: per-rank shard generation
(byte-deterministic row ranges), per-rank partial aggregation (oracle here —
the GPU engine produces the same obx_agg_result by the parity tests), an
all_gather of serialized partial tables, and the exact Python-int merge
(the reference's 2-phase group-by exchange; SURVEY.md §2 collective
inventory: exactness requires gathering partials and adding wide ints
locally, not a 64-bit allreduce).
"""
import multiprocessing as mp
import os

import pytest


def _rank_main(rank, world, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29517"
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch
    import torch.distributed as dist
    dist.init_process_group(backend="gloo", rank=rank, world_size=world)

    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from oceanbase_amd import abi, oracle
    from bench import build_descs, merge_results, result_to_bytes

    rows_per_rank = 7000
    li = oracle.Lineitem(4, rows_per_rank, seed=42, block_bytes=4096,
                         row_base=rank * rows_per_rank)
    filt, agg, n_aggs = build_descs("q1")
    res = oracle.scan_filter_agg(li.bs, filt, agg, nthreads=2)
    blob = result_to_bytes(res)
    t = torch.frombuffer(bytearray(blob), dtype=torch.uint8)
    outs = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(outs, t)
    merged, scanned, passed = merge_results(
        [bytes(o.numpy().tobytes()) for o in outs], n_aggs)
    dist.destroy_process_group()
    q.put((rank, merged, scanned, passed))


def test_two_rank_merge_equals_single_run():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # all ranks agree
    merged0 = results[0][1:]
    for r in results[1:]:
        assert r[1:] == merged0

    # equals a single-process run over the full row range
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from oceanbase_amd import abi, oracle
    from bench import build_descs
    li = oracle.Lineitem(4, 14000, seed=42, block_bytes=4096)
    filt, agg, n_aggs = build_descs("q1")
    res = oracle.scan_filter_agg(li.bs, filt, agg)
    expect = {key: [cnt] + cells
              for key, cnt, cells in abi.result_rows(res, n_aggs)}
    merged, scanned, passed = merged0
    assert scanned == 14000
    assert passed == res.rows_passed
    assert merged == expect


def _rank_main_uneven(rank, world, shards, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29523"
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch
    import torch.distributed as dist
    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from oceanbase_amd import oracle
    from bench import build_descs, merge_results, result_to_bytes

    row_base = sum(shards[:rank])
    li = oracle.Lineitem(4, shards[rank], seed=7, block_bytes=4096,
                         row_base=row_base)
    filt, agg, n_aggs = build_descs("q1")
    res = oracle.scan_filter_agg(li.bs, filt, agg, nthreads=2)
    blob = result_to_bytes(res)
    t = torch.frombuffer(bytearray(blob), dtype=torch.uint8)
    # uneven blobs: exchange sizes first, pad to max (the bench's merge
    # tolerates trailing zero padding via the embedded group count)
    sz = torch.tensor([t.numel()], dtype=torch.int64)
    sizes = [torch.zeros(1, dtype=torch.int64) for _ in range(world)]
    dist.all_gather(sizes, sz)
    mx = int(max(s.item() for s in sizes))
    tp = torch.zeros(mx, dtype=torch.uint8)
    tp[:t.numel()] = t
    outs = [torch.empty_like(tp) for _ in range(world)]
    dist.all_gather(outs, tp)
    merged, scanned, passed = merge_results(
        [bytes(o.numpy().tobytes()[:int(sizes[i].item())])
         for i, o in enumerate(outs)], n_aggs)
    dist.destroy_process_group()
    q.put((rank, merged, scanned, passed))


def test_three_rank_uneven_shards():
    """world_size 3 with unequal shard sizes (the N>1 launch where rows
    don't divide evenly): size-prefixed all_gather with padding, same
    merged table as one process over the union."""
    world = 3
    shards = [5000, 4000, 2800]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main_uneven,
                         args=(r, world, shards, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    merged0 = results[0][1:]
    for r in results[1:]:
        assert r[1:] == merged0

    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from oceanbase_amd import abi, oracle
    from bench import build_descs
    li = oracle.Lineitem(4, sum(shards), seed=7, block_bytes=4096)
    filt, agg, n_aggs = build_descs("q1")
    res = oracle.scan_filter_agg(li.bs, filt, agg)
    expect = {key: [cnt] + cells
              for key, cnt, cells in abi.result_rows(res, n_aggs)}
    merged, scanned, passed = merged0
    assert scanned == sum(shards)
    assert passed == res.rows_passed
    assert merged == expect
