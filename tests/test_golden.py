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


# ---- string/span encoding byte pins (tests/golden/string_encodings.json;
# any codec change that alters these bytes must be deliberate and
# re-recorded — same policy as the lineitem container pins above) ----

_SG = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                  "string_encodings.json")))


def _sg_block(name):
    import numpy as np
    rows = 1024
    rng = np.random.default_rng(_SG["seed"])
    alpha = np.frombuffer(b"ABCDWXYZ", dtype=np.uint8)
    a = rng.choice(alpha, (rows, 4))
    if name == "hex_char4":
        schema = oracle.make_schema([(abi.T_CHAR, 0, 0, 4)])
        return oracle.encode_block(schema, [a.reshape(-1)],
                                   [abi.ENC_HEX], None)
    b = a.copy(); b[:, 0] = ord("Q"); b[:, 2] = ord("R")
    nb = np.zeros((rows + 7) // 8, dtype=np.uint8); nb[0] |= 3
    if name == "sdiff_char4_nulls":
        schema = oracle.make_schema([(abi.T_CHAR, 0, 0, 4)])
        return oracle.encode_block(schema, [b.reshape(-1)],
                                   [abi.ENC_SDIFF], [nb])
    p = np.zeros((rows, 6), dtype=np.uint8)
    prefs = [b"ABC", b"XYZ"]
    for r in range(rows):
        pp = prefs[int(rng.integers(0, 2))]
        p[r] = np.frombuffer(pp + bytes(rng.choice(alpha[:4], 3)),
                             dtype=np.uint8)
    if name == "prefix_char6":
        schema = oracle.make_schema([(abi.T_CHAR, 0, 0, 6)])
        return oracle.encode_block(schema, [p.reshape(-1)],
                                   [abi.ENC_STRING_PREFIX], None)
    base_i = rng.integers(-10**9, 10**9, rows)
    eq = base_i.copy(); eq[::97] += 3
    wide = rng.choice(alpha, (rows, 6))
    sub = wide[:, 1:4].copy(); sub[::89, 0] ^= 1
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8),
                                 (abi.T_INT, 0, 19, 8),
                                 (abi.T_CHAR, 0, 0, 6),
                                 (abi.T_CHAR, 0, 0, 3)])
    return oracle.encode_block(
        schema, [base_i.view(np.uint8), eq.view(np.uint8),
                 wide.reshape(-1), sub.reshape(-1)],
        [abi.ENC_RAW, abi.ENC_COLUMN_EQUAL, abi.ENC_RAW,
         abi.ENC_COLUMN_SUBSTR], None)


@pytest.mark.parametrize("name", sorted(_SG["cases"]))
def test_string_encoding_bytes_pinned(name):
    import numpy as np  # noqa: F401
    blk = _sg_block(name)
    g = _SG["cases"][name]
    assert len(blk) == g["bytes"]
    assert hashlib.sha256(blk).hexdigest() == g["sha256"]
