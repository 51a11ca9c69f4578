"""TEST INFRASTRUCTURE: oracle-backed CS block decode + PAX transcode.

Moved out of the product package (round-1 oceanbase_amd/cs.py): the
PRODUCT CS path is now the GPU-native load-time transform
(oceanbase_amd/csrc/obx_cs_load.cpp + obx_cs_kernels.hip, exposed as
GpuEngine.load_cs) and never touches liboracle.so. This module keeps the
oracle-side CS decode for parity tests: it CDLLs liboracle.so and
re-encodes CS tables as PAX blocks so CPU tests can cross-check the two
formats through the same query pipeline.

Original module docstring follows.

CS (cs_encoding) block loading: the transformer role.

The reference reads CS-format micro-blocks through
ObCSMicroBlockTransformer (storage/blocksstable/cs_encoding/
ob_cs_micro_block_transformer.cpp), which on load transforms the disk
byte stream into the in-memory scan layout ("full transform") before
the vectorized decoders run. This module restates that role for the
obx engine: a CS block is parsed with the oracle's format layer
(oracle/obx_cs_block.c) and transcoded into the PAX micro-block layout
the engine's GPU kernels scan natively, so CS-encoded data flows
through the SAME hot path (filter + aggregate) with no separate device
decoder. A native C++ transcoder inside obx_engine load is the round-2
follow-up; this host-side path is bit-exact by the cross-format
equivalence tests (tests/test_cs_pipeline_equiv.py).

Integer columns (INTEGER / INT_DICT) become 8-byte T_INT PAX columns;
fixed-length string columns (STRING fixed / STR_DICT with a fixed-len
dict) become T_CHAR columns of that length. Var-length strings have no
fixed-cell PAX equivalent and are rejected (the engine's round-1 PAX
scope is fixed-cell columns).
"""
import ctypes as C
import os

import numpy as np

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
from oceanbase_amd import abi, oracle

_lib = C.CDLL(os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "oracle", "liboracle.so"))


class _DictMeta(C.Structure):
    _pack_ = 1
    _fields_ = [("version", C.c_uint8), ("attrs", C.c_uint8),
                ("distinct_val_cnt", C.c_uint32),
                ("ref_row_cnt", C.c_uint32)]


class _ColView(C.Structure):
    _fields_ = [("version", C.c_uint8), ("type", C.c_uint8),
                ("attrs", C.c_uint8), ("obj_type", C.c_uint8),
                ("null_bitmap", C.POINTER(C.c_uint8)),
                ("int_stream", C.POINTER(C.c_uint8)),
                ("int_stream_len", C.c_size_t),
                ("sm_version", C.c_uint8), ("sm_attr", C.c_uint8),
                ("sm_uncompressed_len", C.c_uint32),
                ("sm_fixed_str_len", C.c_uint32),
                ("off_stream", C.POINTER(C.c_uint8)),
                ("off_stream_len", C.c_size_t),
                ("str_data_off", C.c_uint32),
                ("dm", _DictMeta),
                ("ref_stream", C.POINTER(C.c_uint8)),
                ("ref_stream_len", C.c_size_t)]


class _AllColHeader(C.Structure):
    _pack_ = 1
    _fields_ = [("version", C.c_uint8), ("attrs", C.c_uint8),
                ("all_string_data_length", C.c_uint32),
                ("stream_offsets_length", C.c_uint32),
                ("stream_count", C.c_uint16)]


class _BlockView(C.Structure):
    _fields_ = [("buf", C.POINTER(C.c_uint8)), ("len", C.c_size_t),
                ("rows", C.c_uint32), ("ncols", C.c_uint32),
                ("ach", _AllColHeader),
                ("all_string", C.POINTER(C.c_uint8)),
                ("stream_offsets", C.c_uint32 * 96),
                ("stream_count", C.c_uint32),
                ("col", _ColView * 48)]


_lib.obx_cs_block_dec.restype = C.c_int
_lib.obx_cs_block_dec.argtypes = [C.POINTER(C.c_uint8), C.c_size_t,
                                  C.POINTER(_BlockView)]
_lib.obx_cs_block_get_int.restype = C.c_int
_lib.obx_cs_block_get_int.argtypes = [C.POINTER(_BlockView), C.c_uint32,
                                      C.POINTER(C.c_int64),
                                      C.POINTER(C.c_uint8)]
_lib.obx_cs_block_get_str.restype = C.c_int64
_lib.obx_cs_block_get_str.argtypes = [C.POINTER(_BlockView), C.c_uint32,
                                      C.POINTER(C.c_uint8), C.c_size_t,
                                      C.POINTER(C.c_uint32),
                                      C.POINTER(C.c_uint8)]

COL_INTEGER, COL_STRING, COL_INT_DICT, COL_STR_DICT = 0, 1, 2, 3
STR_FIXED_LEN = 0x2


def decode_block(block):
    """Parse one CS block. Returns (rows, cols) where cols is a list of
    dicts: {"kind": "int"|"str", "values": int64 array | list[bytes],
    "nulls": bool array, "fixed_len": int|None}."""
    buf = (C.c_uint8 * len(block)).from_buffer_copy(block)
    v = _BlockView()
    if _lib.obx_cs_block_dec(buf, len(block), C.byref(v)) != 0:
        raise ValueError("not a CS block")
    rows = v.rows
    nbm = (rows + 7) // 8
    out = []
    for c in range(v.ncols):
        t = v.col[c].type
        nb = np.zeros(nbm, dtype=np.uint8)
        if t in (COL_INTEGER, COL_INT_DICT):
            vals = np.zeros(rows, dtype=np.int64)
            if _lib.obx_cs_block_get_int(
                    C.byref(v), c,
                    vals.ctypes.data_as(C.POINTER(C.c_int64)),
                    nb.ctypes.data_as(C.POINTER(C.c_uint8))) != 0:
                raise ValueError(f"column {c}: int decode failed")
            nulls = np.unpackbits(nb, bitorder="little")[:rows].astype(bool)
            out.append(dict(kind="int", values=vals, nulls=nulls,
                            fixed_len=None))
        else:
            if t == COL_STR_DICT:
                # sm covers the DICT bytes; materialized rows need
                # rows x longest value (bounded by the dict total)
                per = (v.col[c].sm_fixed_str_len
                       if v.col[c].sm_attr & STR_FIXED_LEN
                       else v.col[c].sm_uncompressed_len)
                cap = rows * max(per, 1) + 16
            else:
                cap = v.col[c].sm_uncompressed_len + 16
            bout = (C.c_uint8 * cap)()
            lens = np.zeros(rows, dtype=np.uint32)
            total = _lib.obx_cs_block_get_str(
                C.byref(v), c, bout, cap,
                lens.ctypes.data_as(C.POINTER(C.c_uint32)),
                nb.ctypes.data_as(C.POINTER(C.c_uint8)))
            if total < 0:
                raise ValueError(f"column {c}: string decode failed")
            data = bytes(bout[:total])
            vals, pos = [], 0
            for r in range(rows):
                vals.append(data[pos:pos + int(lens[r])])
                pos += int(lens[r])
            nulls = np.unpackbits(nb, bitorder="little")[:rows].astype(bool)
            # fixed length comes from the stream metadata when declared
            # (ObStringStreamMeta fixed_str_len / STR_FIXED_LEN attr) so
            # all-null or all-empty blocks and per-block value skew cannot
            # change the inferred schema; data inference is the fallback
            # for var-stored columns that happen to be uniform.
            fl = None
            if v.col[c].sm_attr & STR_FIXED_LEN and \
                    v.col[c].sm_fixed_str_len > 0:
                fl = int(v.col[c].sm_fixed_str_len)
            else:
                nn = [len(s) for r, s in enumerate(vals) if not nulls[r]]
                if nn and all(l == nn[0] for l in nn):
                    fl = nn[0]
            out.append(dict(kind="str", values=vals, nulls=nulls,
                            fixed_len=fl))
    return rows, out


def to_pax_blocks(cs_blocks, declared_specs=None):
    """Transcode CS blocks (one logical table, identical schemas) into
    (schema, [pax_block_bytes]) ready for the engine's blockset loader.
    The transform is the load-time step; the scan itself runs on the
    engine's native PAX kernels.

    declared_specs: optional schema-level declaration [(obj_type, scale,
    precision, len), ...] used for columns whose block carries no usable
    length (e.g. an all-null string block with no STR_FIXED_LEN attr)."""
    specs = None
    pax = []
    for block in cs_blocks:
        rows, cols = decode_block(block)
        bspecs = []
        for i, col in enumerate(cols):
            if col["kind"] == "int":
                bspecs.append((abi.T_INT, 0, 19, 8))
            else:
                fl = col["fixed_len"]
                if fl is None and declared_specs is not None:
                    fl = declared_specs[i][3]
                if fl is None:
                    raise ValueError(
                        "var-length string column has no fixed-cell PAX "
                        "equivalent (round-1 engine scope); pass "
                        "declared_specs for all-null blocks")
                bspecs.append((abi.T_CHAR, 0, 0, fl))
        if specs is None:
            specs = bspecs
        elif specs != bspecs:
            raise ValueError("CS blocks disagree on schema")
        schema = oracle.make_schema(specs)
        arrays, nulls = [], []
        for i, col in enumerate(cols):
            if col["kind"] == "int":
                arrays.append(col["values"].copy())
            else:
                fl = specs[i][3]
                buf = np.zeros(rows * fl, dtype=np.uint8)
                for r, s in enumerate(col["values"]):
                    if not col["nulls"][r] and s:
                        buf[r * fl:r * fl + len(s)] = np.frombuffer(
                            s, dtype=np.uint8)
                arrays.append(buf)
            if col["nulls"].any():
                nulls.append(np.packbits(col["nulls"].astype(np.uint8),
                                         bitorder="little"))
            else:
                nulls.append(None)
        pax.append(oracle.encode_block(schema, arrays,
                                       [abi.ENC_AUTO] * len(specs), nulls))
    return oracle.make_schema(specs), pax
