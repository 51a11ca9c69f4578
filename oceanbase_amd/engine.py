"""MI355X product engine bindings (oceanbase_amd/libobx.so, HIP/gfx950).

This is the PRODUCT path. It requires the in-tree HIP library; if the library
is missing or no GPU is present, calls raise — there is no CPU fallback (the
oracle is test infrastructure only).
"""
import ctypes as C
import os

from . import abi

_PKG = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_PKG, "libobx.so")

OBX_NO_GPU = -7001


class EngineUnavailable(RuntimeError):
    pass


def _load():
    if not os.path.exists(_SO):
        raise EngineUnavailable(
            f"{_SO} not built — run oceanbase_amd/csrc/build.sh (the product "
            "path has no CPU fallback)")
    lib = C.CDLL(_SO)
    lib.obx_gpu_open.restype = C.c_int
    lib.obx_gpu_open.argtypes = [C.c_int, C.POINTER(C.c_void_p)]
    lib.obx_gpu_close.restype = C.c_int
    lib.obx_gpu_close.argtypes = [C.c_void_p]
    lib.obx_gpu_load_blocks.restype = C.c_int
    lib.obx_gpu_load_blocks.argtypes = [C.c_void_p, C.POINTER(abi.BlockSet)]
    lib.obx_gpu_load_cs_blocks.restype = C.c_int
    lib.obx_gpu_load_cs_blocks.argtypes = [C.c_void_p,
                                           C.POINTER(abi.BlockSet)]
    lib.obx_gpu_free_blocks.restype = C.c_int
    lib.obx_gpu_free_blocks.argtypes = [C.c_void_p, C.c_int]
    lib.obx_gpu_filter.restype = C.c_int
    lib.obx_gpu_filter.argtypes = [C.c_void_p, C.c_int,
                                   C.POINTER(abi.FilterDesc), C.c_int]
    lib.obx_gpu_fetch_bitmap.restype = C.c_int
    lib.obx_gpu_fetch_bitmap.argtypes = [C.c_void_p, C.c_int, C.c_void_p,
                                         C.c_int64]
    lib.obx_gpu_fetch_row_ids.restype = C.c_int
    lib.obx_gpu_fetch_row_ids.argtypes = [C.c_void_p, C.c_int, C.c_void_p,
                                          C.c_int64, C.POINTER(C.c_uint64)]
    lib.obx_gpu_fetch_blk_counts.restype = C.c_int
    lib.obx_gpu_fetch_blk_counts.argtypes = [C.c_void_p, C.c_int, C.c_void_p,
                                             C.c_int64]
    lib.obx_gpu_decode.restype = C.c_int
    lib.obx_gpu_decode.argtypes = [C.c_void_p, C.c_int,
                                   C.POINTER(C.c_uint16), C.c_uint16]
    lib.obx_gpu_fetch_col.restype = C.c_int
    lib.obx_gpu_fetch_col.argtypes = [C.c_void_p, C.c_int, C.c_uint16,
                                      C.c_void_p, C.c_int64]
    lib.obx_gpu_scan_filter_agg.restype = C.c_int
    lib.obx_gpu_scan_filter_agg.argtypes = [C.c_void_p, C.c_int,
                                            C.POINTER(abi.FilterDesc),
                                            C.POINTER(abi.AggDesc),
                                            C.POINTER(abi.AggResult)]
    lib.obx_gpu_last_kernel_ms.restype = C.c_double
    lib.obx_gpu_last_kernel_ms.argtypes = [C.c_void_p]
    lib.obx_gpu_last_prep_ms.restype = C.c_double
    lib.obx_gpu_last_prep_ms.argtypes = [C.c_void_p]
    lib.obx_gpu_last_jit.restype = C.c_int
    lib.obx_gpu_last_jit.argtypes = [C.c_void_p]
    for f in ("obx_gpu_total_rows", "obx_gpu_total_bytes",
              "obx_gpu_last_survivors"):
        getattr(lib, f).restype = C.c_uint64
        getattr(lib, f).argtypes = [C.c_void_p, C.c_int]
    return lib


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = _load()
    return _lib


class GpuEngine:
    def __init__(self, device=0):
        self._lib = lib()
        self._ctx = C.c_void_p()
        rc = self._lib.obx_gpu_open(device, C.byref(self._ctx))
        if rc == OBX_NO_GPU:
            raise EngineUnavailable("no HIP device visible (product path "
                                    "refuses to run without a GPU)")
        if rc != 0:
            raise RuntimeError(f"obx_gpu_open failed: {rc}")

    def close(self):
        if self._ctx:
            self._lib.obx_gpu_close(self._ctx)
            self._ctx = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def load(self, bs):
        h = self._lib.obx_gpu_load_blocks(self._ctx, C.byref(bs))
        if h < 0:
            raise RuntimeError(f"obx_gpu_load_blocks failed: {h}")
        return h

    def load_cs(self, cs_blocks, schema):
        """GPU-native CS load: device kernels decode the CS streams into
        the engine's scan layout (ObCSMicroBlockTransformer position)."""
        from . import cs as _cs
        bs, keep = _cs.make_blockset(cs_blocks, schema)
        h = self._lib.obx_gpu_load_cs_blocks(self._ctx, C.byref(bs))
        if h < 0:
            raise RuntimeError(f"obx_gpu_load_cs_blocks failed: {h}")
        return h

    def free(self, handle):
        self._lib.obx_gpu_free_blocks(self._ctx, handle)

    def filter(self, handle, filter_desc, want_row_ids=False):
        rc = self._lib.obx_gpu_filter(self._ctx, handle,
                                      C.byref(filter_desc),
                                      1 if want_row_ids else 0)
        if rc != 0:
            raise RuntimeError(f"obx_gpu_filter failed: {rc}")
        return self._lib.obx_gpu_last_survivors(self._ctx, handle)

    def fetch_bitmap(self, handle):
        import numpy as np
        rows = self._lib.obx_gpu_total_rows(self._ctx, handle)
        out = np.zeros((rows + 7) // 8, dtype=np.uint8)
        rc = self._lib.obx_gpu_fetch_bitmap(self._ctx, handle,
                                            out.ctypes.data_as(C.c_void_p),
                                            out.nbytes)
        if rc != 0:
            raise RuntimeError(f"fetch_bitmap failed: {rc}")
        return out

    def fetch_row_ids(self, handle):
        import numpy as np
        rows = self._lib.obx_gpu_total_rows(self._ctx, handle)
        out = np.zeros(rows, dtype=np.int32)
        n = C.c_uint64(0)
        rc = self._lib.obx_gpu_fetch_row_ids(self._ctx, handle,
                                             out.ctypes.data_as(C.c_void_p),
                                             rows, C.byref(n))
        if rc != 0:
            raise RuntimeError(f"fetch_row_ids failed: {rc}")
        return out, int(n.value)

    def fetch_blk_counts(self, handle, n_blocks):
        import numpy as np
        out = np.zeros(n_blocks, dtype=np.uint32)
        rc = self._lib.obx_gpu_fetch_blk_counts(
            self._ctx, handle, out.ctypes.data_as(C.c_void_p), n_blocks)
        if rc != 0:
            raise RuntimeError(f"fetch_blk_counts failed: {rc}")
        return out

    def decode(self, handle, cols):
        arr = (C.c_uint16 * len(cols))(*cols)
        rc = self._lib.obx_gpu_decode(self._ctx, handle, arr, len(cols))
        if rc != 0:
            raise RuntimeError(f"obx_gpu_decode failed: {rc}")

    def fetch_col(self, handle, col, datum_len):
        import numpy as np
        rows = self._lib.obx_gpu_total_rows(self._ctx, handle)
        out = np.zeros(rows * datum_len, dtype=np.uint8)
        rc = self._lib.obx_gpu_fetch_col(self._ctx, handle, col,
                                         out.ctypes.data_as(C.c_void_p),
                                         out.nbytes)
        if rc != 0:
            raise RuntimeError(f"fetch_col failed: {rc}")
        return out

    def scan_filter_agg(self, handle, filter_desc, agg_desc):
        res = abi.AggResult()
        rc = self._lib.obx_gpu_scan_filter_agg(
            self._ctx, handle,
            C.byref(filter_desc) if filter_desc is not None else None,
            C.byref(agg_desc) if agg_desc is not None else None,
            C.byref(res))
        if rc != 0:
            raise RuntimeError(f"obx_gpu_scan_filter_agg failed: {rc}")
        return res

    def scan_filter_agg_paged(self, handle, filter_desc, agg_desc):
        """Scan allowing > OBX_MAX_GROUPS groups: returns
        (AggResult, [GroupRow...]) with the FULL sorted row list fetched
        through the pagination surface (obx_gpu_agg_fetch)."""
        res = abi.AggResult()
        rc = self._lib.obx_gpu_scan_filter_agg(
            self._ctx, handle,
            C.byref(filter_desc) if filter_desc is not None else None,
            C.byref(agg_desc) if agg_desc is not None else None,
            C.byref(res))
        if rc not in (0, abi.OBX_BUF_NOT_ENOUGH):
            raise RuntimeError(f"obx_gpu_scan_filter_agg failed: {rc}")
        return res, self.agg_fetch_all(handle)

    def agg_fetch_all(self, handle, page=256):
        rows, start = [], 0
        buf = (abi.GroupRow * page)()
        n_out = C.c_uint32()
        total = C.c_uint64()
        while True:
            rc = self._lib.obx_gpu_agg_fetch(self._ctx, handle, start, page,
                                             buf, C.byref(n_out),
                                             C.byref(total))
            if rc != 0:
                raise RuntimeError(f"obx_gpu_agg_fetch failed: {rc}")
            for i in range(n_out.value):
                rows.append(abi.GroupRow.from_buffer_copy(buf[i]))
            start += n_out.value
            if start >= total.value or n_out.value == 0:
                break
        return rows

    def last_kernel_ms(self):
        return float(self._lib.obx_gpu_last_kernel_ms(self._ctx))

    def last_jit(self):
        """True if the last scan ran the hipRTC plan-specialized kernel."""
        return bool(self._lib.obx_gpu_last_jit(self._ctx))

    def last_prep_ms(self):
        """Device time of the last query's prep (plan upload + per-block
        filter lowering) — per-operator monitoring, SURVEY §5."""
        return float(self._lib.obx_gpu_last_prep_ms(self._ctx))

    def total_bytes(self, handle):
        return int(self._lib.obx_gpu_total_bytes(self._ctx, handle))

    def total_rows(self, handle):
        return int(self._lib.obx_gpu_total_rows(self._ctx, handle))
