"""Committed golden fixtures (tests/golden/lineitem_small.json).

The reference ships no stored golden vectors for this path,
so these are OURS: oracle-generated container digests + exact query results
for small seeded workloads, committed to pin the byte format and the
aggregate arithmetic across rounds (any codec or generator change that
alters bytes or results must be deliberate and re-recorded).
"""
import hashlib
import json
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oceanbase_amd import abi, oracle  # noqa: E402
from bench import build_descs  # noqa: E402

GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "lineitem_small.json")))

CASES = [("config2", 2, "filter-int64"), ("config3", 3, "decode-filter"),
         ("q1", 4, "q1"), ("q6", 6, "q6")]


@pytest.mark.parametrize("name,config,wl", CASES)
def test_container_bytes_pinned(name, config, wl):
    g = GOLDEN[name]
    li = oracle.Lineitem(config, g["rows"], seed=GOLDEN["seed"],
                         block_bytes=GOLDEN["block_bytes"])
    assert li.n_blocks == g["n_blocks"]
    assert li.total_bytes == g["total_bytes"]
    h = hashlib.sha256()
    for b in range(li.n_blocks):
        h.update(li.block(b))
    assert h.hexdigest() == g["container_sha256"]


@pytest.mark.parametrize("name,config,wl", CASES)
def test_results_pinned(name, config, wl):
    g = GOLDEN[name]
    li = oracle.Lineitem(config, g["rows"], seed=GOLDEN["seed"],
                         block_bytes=GOLDEN["block_bytes"])
    filt, agg, n_aggs = build_descs(wl)
    res = oracle.scan_filter_agg(li.bs, filt, agg)
    assert res.rows_passed == g["rows_passed"]
    if agg is not None:
        got = [{"key": key.hex(), "count": rc,
                "cells": [str(c) for c in cells]}
               for key, rc, cells in abi.result_rows(res, n_aggs)]
        assert got == g["groups"]


@pytest.mark.gpu
@pytest.mark.parametrize("name,config,wl", CASES)
def test_gpu_matches_golden(name, config, wl):
    from oceanbase_amd.engine import GpuEngine
    g = GOLDEN[name]
    li = oracle.Lineitem(config, g["rows"], seed=GOLDEN["seed"],
                         block_bytes=GOLDEN["block_bytes"])
    eng = GpuEngine(0)
    h = eng.load(li.bs)
    filt, agg, n_aggs = build_descs(wl)
    if agg is None:
        survivors = eng.filter(h, filt)
        assert survivors == g["rows_passed"]
    else:
        res = eng.scan_filter_agg(h, filt, agg)
        assert res.rows_passed == g["rows_passed"]
        got = [{"key": key.hex(), "count": rc,
                "cells": [str(c) for c in cells]}
               for key, rc, cells in abi.result_rows(res, n_aggs)]
        assert got == g["groups"]
    eng.close()
