"""CS (cs_encoding) block loading — the PRODUCT path.

The reference reads CS-format micro blocks through
ObCSMicroBlockTransformer (storage/blocksstable/cs_encoding/
ob_cs_micro_block_transformer.cpp): a load-time transform from the disk
stream format into the in-memory scan layout. The engine's equivalent is
GPU-native: `load_cs` hands the raw CS blocks to libobx.so
(obx_gpu_load_cs_blocks), whose host side parses stream metadata and
whose device kernels (obx_cs_kernels.hip) decode every stream — RAW
widths, the PFoR/RLE codec families, dict refs, null recovery — straight
into an HBM arena in the engine's native scan layout. The returned
handle scans/filters/aggregates exactly like a PAX handle.

liboracle.so is never touched here; the oracle-side CS decode lives in
tests/cs_oracle_util.py (test infrastructure) and parity tests compare
the two bit-exactly.
"""
import ctypes as C

import numpy as np

from . import abi


def make_blockset(blocks, schema):
    """Build an obx_blockset over concatenated CS block bytes.

    blocks: list of bytes-like CS blocks; schema: abi schema array
    (make_schema). Returns (blockset, keepalive)."""
    offs = np.zeros(len(blocks) + 1, dtype=np.uint64)
    total = 0
    for i, b in enumerate(blocks):
        offs[i] = total
        total += len(b)
    offs[len(blocks)] = total
    data = np.zeros(total, dtype=np.uint8)
    for i, b in enumerate(blocks):
        data[int(offs[i]):int(offs[i]) + len(b)] = np.frombuffer(
            bytes(b), dtype=np.uint8)
    bs = abi.BlockSet()
    bs.data = data.ctypes.data_as(C.POINTER(C.c_uint8))
    bs.block_offsets = offs.ctypes.data_as(C.POINTER(C.c_uint64))
    bs.n_blocks = len(blocks)
    bs.n_cols = len(schema)
    bs.cols = schema
    bs.total_rows = 0  # filled by the loader from block headers
    return bs, (data, offs, schema)
