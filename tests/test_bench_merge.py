"""Unit tests for bench.py's cross-rank result merge (the distributed
2-phase group-by merge: sum-family adds, MIN/MAX folds)."""
import ctypes as C
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pytest

from oceanbase_amd import abi
import bench


def _blob(key, cnt, cells):
    r = abi.AggResult()
    r.n_groups = 1
    r.rows_scanned = cnt
    r.rows_passed = cnt
    g = r.groups[0]
    g.key[0] = key
    g.key_len = 1
    g.row_count = cnt
    for i, v in enumerate(cells):
        g.cells[i].limb[0] = v & ((1 << 64) - 1)
        s = ~0 if v < 0 else 0
        for l in (1, 2, 3):
            g.cells[i].limb[l] = s & ((1 << 64) - 1)
    return bytes(r)


def test_merge_sums_and_minmax():
    kinds = [abi.AGG_SUM, abi.AGG_MIN, abi.AGG_MAX]
    b1 = _blob(65, 10, [100, -5, 7])
    b2 = _blob(65, 3, [-40, -9, 2])
    groups, scanned, passed = bench.merge_results([b1, b2], 3, kinds)
    assert scanned == 13 and passed == 13
    (key, (cnt, s, mn, mx)), = [(k, tuple(v)) for k, v in groups.items()]
    assert key == b"A" and cnt == 13
    assert s == 60 and mn == -9 and mx == 7


def test_merge_rejects_unmergeable_kind():
    blob = _blob(66, 1, [1])
    with pytest.raises(AssertionError):
        bench.merge_results([blob], 1, [99])
