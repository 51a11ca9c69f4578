"""CPU oracle bindings (oracle/liboracle.so).

TEST INFRASTRUCTURE + the reported CPU baseline of bench.py — never the
product path. Only tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg may import this module (enforced by review, stated in
DESIGN.md; the product engine raises if its HIP library is absent instead of
falling back here).
"""
import ctypes as C
import os
import subprocess

import numpy as np

from . import abi

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_SO = os.path.join(_REPO, "oracle", "liboracle.so")


def _build_if_needed():
    srcs = [os.path.join(_REPO, "oracle", f)
            for f in ("obx_codec.c", "obx_agg.c", "obx_gen.c", "obx_cs.c",
                      "obx_cs_block.c", "obx_cs_block.h",
                      "obx_format.h", "obx_cs.h")]
    if os.path.exists(_SO) and all(
            os.path.getmtime(_SO) >= os.path.getmtime(s) for s in srcs):
        return
    subprocess.run(["make", "-C", os.path.join(_REPO, "oracle")], check=True,
                   capture_output=True)


_build_if_needed()
_lib = C.CDLL(_SO)

_lib.obx_encode_block.restype = C.c_int64
_lib.obx_encode_block.argtypes = [
    C.POINTER(abi.ColSchema), C.c_uint16, C.POINTER(C.c_void_p),
    C.POINTER(C.c_void_p), C.c_uint32, C.POINTER(C.c_uint8),
    C.c_void_p, C.c_int64]
_lib.obx_decode_block.restype = C.c_int
_lib.obx_decode_block.argtypes = [
    C.POINTER(abi.ColSchema), C.c_uint16, C.c_void_p, C.c_int64,
    C.POINTER(C.c_uint16), C.c_uint16, C.POINTER(C.c_void_p),
    C.POINTER(C.c_void_p), C.POINTER(C.c_uint32)]
_lib.obx_cpu_filter_block.restype = C.c_int
_lib.obx_cpu_filter_block.argtypes = [
    C.POINTER(abi.ColSchema), C.c_uint16, C.c_void_p, C.c_int64,
    C.POINTER(abi.FilterDesc), C.c_void_p, C.POINTER(C.c_uint32),
    C.POINTER(C.c_uint32)]
_lib.obx_cpu_scan_filter_agg.restype = C.c_int
_lib.obx_cpu_scan_filter_agg.argtypes = [
    C.POINTER(abi.BlockSet), C.POINTER(abi.FilterDesc),
    C.POINTER(abi.AggDesc), C.c_int, C.POINTER(abi.AggResult)]
_lib.obx_gen_lineitem.restype = C.c_int64
_lib.obx_gen_lineitem.argtypes = [
    C.c_int, C.c_uint64, C.c_uint64, C.c_uint32, C.c_uint64,
    C.POINTER(C.POINTER(C.c_uint8)), C.POINTER(C.POINTER(C.c_uint64)),
    C.POINTER(abi.ColSchema), C.POINTER(C.c_uint16)]
_lib.obx_date_days.restype = C.c_int64
_lib.obx_date_days.argtypes = [C.c_int, C.c_int, C.c_int]

_libc = C.CDLL(None)
_libc.free.argtypes = [C.c_void_p]


def date_days(y, m, d):
    return _lib.obx_date_days(y, m, d)


def make_schema(cols):
    """cols: list of (obj_type, scale, precision, len)."""
    arr = (abi.ColSchema * len(cols))()
    for i, (t, s, p, l) in enumerate(cols):
        arr[i].obj_type, arr[i].scale, arr[i].precision, arr[i].len = t, s, p, l
    return arr


def encode_block(schema, col_arrays, encodings, null_bitmaps=None):
    """schema: (ColSchema*N); col_arrays: list of numpy arrays (bytes laid out
    as row_count*len per column); encodings: list of abi.ENC_*.
    Returns the encoded block as bytes."""
    n_cols = len(col_arrays)
    rows = None
    ptrs = (C.c_void_p * n_cols)()
    for i, a in enumerate(col_arrays):
        a = np.ascontiguousarray(a)
        col_arrays[i] = a
        r = a.nbytes // schema[i].len
        assert rows is None or rows == r
        rows = r
        ptrs[i] = a.ctypes.data_as(C.c_void_p).value
    nulls = None
    if null_bitmaps is not None:
        nulls = (C.c_void_p * n_cols)()
        for i, nb in enumerate(null_bitmaps):
            nulls[i] = (None if nb is None
                        else np.ascontiguousarray(nb).ctypes.data_as(C.c_void_p).value)
    encs = (C.c_uint8 * n_cols)(*encodings)
    cap = 4096
    for i in range(n_cols):
        # worst case: forced RLE/DICT on high-cardinality data stores the
        # runs/dict alongside full-width entries
        cap += rows * (3 * schema[i].len + 10) + 4096
    out = np.zeros(cap, dtype=np.uint8)
    sz = _lib.obx_encode_block(schema, n_cols, ptrs, nulls, rows, encs,
                               out.ctypes.data_as(C.c_void_p), cap)
    if sz < 0:
        raise RuntimeError(f"obx_encode_block failed: {sz}")
    return bytes(out[:sz]) + b"\x00" * 16  # bitstream read slack


def decode_block(schema, n_cols, block, proj_cols):
    """Returns (row_count, [np.uint8 arrays of rows*len per projected col],
    [null bitmap bytes or None])."""
    blk = np.frombuffer(block, dtype=np.uint8)
    rc = C.c_uint32(0)
    # first call with tiny probe to learn row count: header row_count at 16..20
    rows = int(np.frombuffer(block[16:20], dtype=np.uint32)[0])
    outs, nulls_out = [], []
    optr = (C.c_void_p * len(proj_cols))()
    nptr = (C.c_void_p * len(proj_cols))()
    for i, c in enumerate(proj_cols):
        a = np.zeros(rows * schema[c].len, dtype=np.uint8)
        nb = np.zeros((rows + 7) // 8, dtype=np.uint8)
        outs.append(a)
        nulls_out.append(nb)
        optr[i] = a.ctypes.data_as(C.c_void_p).value
        nptr[i] = nb.ctypes.data_as(C.c_void_p).value
    proj = (C.c_uint16 * len(proj_cols))(*proj_cols)
    st = _lib.obx_decode_block(schema, n_cols, blk.ctypes.data_as(C.c_void_p),
                               len(block), proj, len(proj_cols), optr, nptr,
                               C.byref(rc))
    if st != abi.OBX_SUCCESS:
        raise RuntimeError(f"obx_decode_block failed: {st}")
    return rows, outs, nulls_out


def filter_block(schema, n_cols, block, filter_desc):
    blk = np.frombuffer(block, dtype=np.uint8)
    rows = int(np.frombuffer(block[16:20], dtype=np.uint32)[0])
    bits = np.zeros((rows + 7) // 8 + 8, dtype=np.uint8)
    rc, pc = C.c_uint32(0), C.c_uint32(0)
    st = _lib.obx_cpu_filter_block(schema, n_cols,
                                   blk.ctypes.data_as(C.c_void_p), len(block),
                                   C.byref(filter_desc),
                                   bits.ctypes.data_as(C.c_void_p),
                                   C.byref(rc), C.byref(pc))
    if st != abi.OBX_SUCCESS:
        raise RuntimeError(f"obx_cpu_filter_block failed: {st}")
    return bits[: (rows + 7) // 8], int(pc.value)


class Lineitem:
    """Generated blockset, owned C memory."""

    def __init__(self, config, rows, seed=42, block_bytes=16384, row_base=0):
        self._data = C.POINTER(C.c_uint8)()
        self._offs = C.POINTER(C.c_uint64)()
        cols = (abi.ColSchema * 8)()
        ncols = C.c_uint16(0)
        nb = _lib.obx_gen_lineitem(config, rows, seed, block_bytes, row_base,
                                   C.byref(self._data), C.byref(self._offs),
                                   cols, C.byref(ncols))
        if nb < 0:
            raise RuntimeError(f"obx_gen_lineitem failed: {nb}")
        self.n_blocks = int(nb)
        self.n_cols = int(ncols.value)
        self.schema = (abi.ColSchema * self.n_cols)(*cols[: self.n_cols])
        self.total_rows = rows
        self.total_bytes = int(self._offs[self.n_blocks])
        self.bs = abi.BlockSet()
        self.bs.data = self._data
        self.bs.block_offsets = self._offs
        self.bs.n_blocks = self.n_blocks
        self.bs.n_cols = self.n_cols
        self.bs.cols = C.cast(self.schema, C.POINTER(abi.ColSchema))
        self.bs.total_rows = rows

    def block(self, i):
        off = self._offs[i]
        end = self._offs[i + 1]
        return C.string_at(C.addressof(self._data.contents) + off,
                           end - off) + b"\x00" * 16

    def subset(self, n_blocks):
        """BlockSet view over the first n_blocks (for bounded CPU baseline)."""
        sub = abi.BlockSet()
        sub.data = self._data
        sub.block_offsets = self._offs
        sub.n_blocks = min(n_blocks, self.n_blocks)
        sub.n_cols = self.n_cols
        sub.cols = C.cast(self.schema, C.POINTER(abi.ColSchema))
        rows = 0
        sub.total_rows = 0  # caller may not need it
        return sub

    def __del__(self):
        try:
            if self._data:
                _libc.free(self._data)
            if self._offs:
                _libc.free(self._offs)
        except Exception:
            pass


def scan_filter_agg(bs, filter_desc, agg_desc, nthreads=0):
    res = abi.AggResult()
    st = _lib.obx_cpu_scan_filter_agg(
        C.byref(bs), C.byref(filter_desc) if filter_desc is not None else None,
        C.byref(agg_desc) if agg_desc is not None else None, nthreads,
        C.byref(res))
    if st != abi.OBX_SUCCESS:
        raise RuntimeError(f"obx_cpu_scan_filter_agg failed: {st}")
    return res


def scan_filter_agg_paged(bs, filter_desc, agg_desc, nthreads=0):
    """Scan allowing > OBX_MAX_GROUPS groups; returns (AggResult, rows)."""
    res = abi.AggResult()
    st = _lib.obx_cpu_scan_filter_agg(
        C.byref(bs), C.byref(filter_desc) if filter_desc is not None else None,
        C.byref(agg_desc) if agg_desc is not None else None, nthreads,
        C.byref(res))
    if st not in (abi.OBX_SUCCESS, abi.OBX_BUF_NOT_ENOUGH):
        raise RuntimeError(f"obx_cpu_scan_filter_agg failed: {st}")
    rows, start, page = [], 0, 256
    buf = (abi.GroupRow * page)()
    n_out = C.c_uint32()
    total = C.c_uint64()
    while True:
        rc = _lib.obx_cpu_agg_fetch(start, page, buf, C.byref(n_out),
                                    C.byref(total))
        if rc != 0:
            raise RuntimeError(f"obx_cpu_agg_fetch failed: {rc}")
        for i in range(n_out.value):
            rows.append(abi.GroupRow.from_buffer_copy(buf[i]))
        start += n_out.value
        if start >= total.value or n_out.value == 0:
            break
    return res, rows
