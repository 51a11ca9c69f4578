"""CPU checks of the hipRTC query-codegen source (no GPU needed:
codegen and strategy selection are host code)."""
import ctypes as C
import os

from oceanbase_amd import abi


def test_filter_jit_source_strip_mined():
    """The bitmap-filter JIT strip-mines R rows/lane with one packed read
    (R*W<=64) and assembles the bitmap via wave-synchronous LDS ORs."""
    import ctypes as C
    from oceanbase_amd import abi
    lib = C.CDLL(os.path.join(os.path.dirname(abi.__file__), "libobx.so"))
    filt = abi.make_filter([dict(col=0, op=abi.OP_LE, lo=10561)])
    n = 4
    cols = (abi.ColSchema * n)()
    for c in range(n):
        cols[c].obj_type = abi.T_INT
    flags = (C.c_uint8 * n)(*[16] * n)      # rangefam
    cmin = (C.c_int64 * n)()
    cmax = (C.c_int64 * n)(*[30000] * n)
    maxcnt = (C.c_uint32 * n)()
    maxw = (C.c_uint32 * n)(*[13] * n)       # 13-bit packed -> R=4
    buf = C.create_string_buffer(1 << 20)
    lib.obx_jit_dump_src.restype = C.c_int64
    sz = lib.obx_jit_dump_src(C.byref(filt), None, cols, n, flags, cmin,
                              cmax, maxcnt, maxw, 5460, 0, buf, len(buf))
    assert sz > 0
    src = buf.raw[:sz].decode()
    # narrow stream, fits LDS budget -> staged mode (stream DMA'd to LDS)
    assert "__shared__ uint8_t seg[" in src
    assert "global_load_lds" in src
    # forced direct mode -> strip-mined loads + LDS mask assembly
    os.environ["OBX_JIT_FSTAGE"] = "0"
    try:
        sz = lib.obx_jit_dump_src(C.byref(filt), None, cols, n, flags,
                                  cmin, cmax, maxcnt, maxw, 5460, 0, buf,
                                  len(buf))
        assert sz > 0
        src = buf.raw[:sz].decode()
        assert "#define FR 4" in src
        assert "fm[wv][lane >> 4]" in src   # LDS mask assembly
        assert "FR * l0W" in src            # one packed read per R rows
    finally:
        del os.environ["OBX_JIT_FSTAGE"]
    # wide column (64-bit raw): no strip mining, ballot path
    maxw2 = (C.c_uint32 * n)(*[64] * n)
    sz = lib.obx_jit_dump_src(C.byref(filt), None, cols, n, flags, cmin,
                              cmax, maxcnt, maxw2, 2048, 0, buf, len(buf))
    assert sz > 0
    src = buf.raw[:sz].decode()
    assert "__ballot" in src


def test_v2_scan_source_min_max():
    """MIN/MAX aggregates are v2-eligible: CTX columns via LDS CAS cells
    (wmm), dict columns read off the histogram at flush."""
    from oceanbase_amd import oracle
    lib = C.CDLL(os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "oceanbase_amd", "libobx.so"))
    filt = abi.make_filter([dict(col=6, op=abi.OP_LE,
                                 lo=oracle.date_days(1998, 9, 2))])
    aggs = [dict(kind=abi.AGG_COUNT),
            dict(kind=abi.AGG_SUM, col_a=1),
            dict(kind=abi.AGG_MIN, col_a=1),    # CTX (raw8) -> wmm
            dict(kind=abi.AGG_MAX, col_a=0)]    # dict col -> hist flush
    agg = abi.make_agg([4, 5], aggs)
    n = 7
    cols = (abi.ColSchema * n)()
    for c in range(n):
        cols[c].obj_type = abi.T_INT
    flags = (C.c_uint8 * n)(*[37, 12, 37, 37, 37, 37, 20])
    cmin = (C.c_int64 * n)(*[0, 90000, 0, 0, 0, 0, 8000])
    cmax = (C.c_int64 * n)(*[50, 10500000, 10, 8, 2, 1, 11000])
    maxcnt = (C.c_uint32 * n)(*[50, 0, 11, 9, 3, 2, 0])
    maxw = (C.c_uint32 * n)(*[6, 64, 4, 4, 2, 1, 13])
    buf = C.create_string_buffer(1 << 20)
    lib.obx_jit_dump_src.restype = C.c_int64
    sz = lib.obx_jit_dump_src(C.byref(filt), C.byref(agg), cols, n, flags,
                              cmin, cmax, maxcnt, maxw, 1365, 0, buf,
                              len(buf))
    assert sz > 0
    src = buf.raw[:sz].decode()
    assert "k_jit_scan" in src
    assert "wmm[" in src                    # CAS min cells (CTX MIN)
    assert "cas_minmax" in src
    assert "INT64_MAX" in src               # MIN sentinel init


def test_filter_jit_folds_or_program():
    """AND/OR combine programs compile to one closed-form expression in
    the staged filter JIT (block verdict in three-valued logic, row eval
    as booleans)."""
    lib = C.CDLL(os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "oceanbase_amd", "libobx.so"))
    filt = abi.make_filter(
        [dict(col=0, op=abi.OP_LE, lo=10561),
         dict(col=1, op=abi.OP_GT, lo=5),
         dict(col=2, op=abi.OP_EQ, lo=3)],
        prog=[0, 1, abi.TOK_AND, 2, abi.TOK_OR])
    n = 4
    cols = (abi.ColSchema * n)()
    for c in range(n):
        cols[c].obj_type = abi.T_INT
    flags = (C.c_uint8 * n)(*[16] * n)
    cmin = (C.c_int64 * n)()
    cmax = (C.c_int64 * n)(*[30000] * n)
    maxcnt = (C.c_uint32 * n)()
    maxw = (C.c_uint32 * n)(*[13, 8, 6, 20])
    buf = C.create_string_buffer(1 << 20)
    lib.obx_jit_dump_src.restype = C.c_int64
    sz = lib.obx_jit_dump_src(C.byref(filt), None, cols, n, flags, cmin,
                              cmax, maxcnt, maxw, 4000, 0, buf, len(buf))
    assert sz > 0
    src = buf.raw[:sz].decode()
    assert "f3or(f3and(c0, c1), c2)" in src      # block verdict fold
    assert "(lf0 && lf1) || lf2" in src          # row fold
