"""Property-based roundtrip pinning (hypothesis): any fixed-datum column
under any eligible encoding must decode back bit-exactly through BOTH the
oracle and the independent Python format model (the
reference's own encoder tests are round-trip property tests —
test_column_decoder.h style — so ours are too)."""
import os
import sys

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from oceanbase_amd import abi, oracle  # noqa: E402
import pymodel  # noqa: E402


@settings(max_examples=40, deadline=None)
@given(st.data())
def test_int_column_roundtrip(data):
    rows = data.draw(st.integers(16, 600))
    enc = data.draw(st.sampled_from(
        [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_RLE, abi.ENC_INT_DIFF,
         abi.ENC_AUTO]))
    lo = data.draw(st.integers(-2**62, 2**62 - 1))
    span = data.draw(st.integers(1, 10**6))
    rng = np.random.default_rng(data.draw(st.integers(0, 2**31)))
    v = rng.integers(lo, lo + span, rows, dtype=np.int64)
    nulls = None
    if data.draw(st.booleans()):
        nulls = np.zeros((rows + 7) // 8, dtype=np.uint8)
        for r in rng.choice(rows, max(1, rows // 10), replace=False):
            nulls[r // 8] |= 1 << (r % 8)
    schema = oracle.make_schema([(abi.T_INT, 0, 19, 8)])
    blk = oracle.encode_block(schema, [v.view(np.uint8)], [enc], [nulls])
    rc, outs, nbs = oracle.decode_block(schema, 1, blk, [0])
    got = np.frombuffer(outs[0], dtype=np.int64)
    vals = pymodel.Block(blk, [(abi.T_INT, 0, 19, 8)]).decode_col(0)
    for r in range(rows):
        isn = (nbs[0][r // 8] >> (r % 8)) & 1
        wn = bool(nulls is not None and (nulls[r // 8] >> (r % 8)) & 1)
        assert bool(isn) == wn
        assert (vals[r] is None) == wn
        if not wn:
            assert got[r] == v[r] == vals[r]


@settings(max_examples=30, deadline=None)
@given(st.data())
def test_char_column_roundtrip(data):
    rows = data.draw(st.integers(16, 400))
    length = data.draw(st.sampled_from([1, 2, 4, 8]))
    enc = data.draw(st.sampled_from(
        [abi.ENC_RAW, abi.ENC_DICT, abi.ENC_SDIFF, abi.ENC_HEX,
         abi.ENC_AUTO]))
    rng = np.random.default_rng(data.draw(st.integers(0, 2**31)))
    alpha = np.frombuffer(b"ABCDWXYZ", dtype=np.uint8)
    a = rng.choice(alpha, (rows, length))
    if data.draw(st.booleans()):  # common positions (string-diff shape)
        a[:, 0] = ord("Q")
    schema = oracle.make_schema([(abi.T_CHAR, 0, 0, length)])
    try:
        blk = oracle.encode_block(schema, [a.reshape(-1)], [enc], None)
    except RuntimeError:
        return  # encoder declined (eligibility) — allowed
    rc, outs, nbs = oracle.decode_block(schema, 1, blk, [0])
    got = np.frombuffer(outs[0], dtype=np.uint8).reshape(rows, length)
    vals = pymodel.Block(blk, [(abi.T_CHAR, 0, 0, length)]).decode_col(0)
    for r in range(rows):
        assert (got[r] == a[r]).all()
        assert vals[r] == int.from_bytes(a[r].tobytes(), "little")
