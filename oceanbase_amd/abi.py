"""ctypes mirror of include/obx.h — the C-ABI drop-in boundary.

Struct layouts must match include/obx.h exactly (which in turn cites the
reference structs it mirrors; see that header).
"""
import ctypes as C

# ---- enums -----------------------------------------------------------------
ENC_RAW, ENC_DICT, ENC_RLE, ENC_CONST, ENC_INT_DIFF, ENC_SDIFF, ENC_HEX, ENC_STRING_PREFIX, ENC_COLUMN_EQUAL, ENC_COLUMN_SUBSTR, ENC_AUTO = 0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 255

T_INT, T_INT32, T_DATE, T_CHAR, T_DECIMAL_INT = 5, 4, 19, 23, 50

OP_EQ, OP_LE, OP_LT, OP_GE, OP_GT, OP_NE, OP_BT, OP_IN, OP_NU, OP_NN = range(10)
OP_BLACK = 10

AGG_COUNT, AGG_SUM, AGG_MIN, AGG_MAX, AGG_SUM_PROD2, AGG_SUM_PROD3, AGG_SUM_MUL = range(7)

OBX_SUCCESS = 0
OBX_BUF_NOT_ENOUGH = -4009

MAX_GROUPS = 64
MAX_KEY_BYTES = 16


class ColSchema(C.Structure):
    _fields_ = [("obj_type", C.c_uint8), ("scale", C.c_int8),
                ("precision", C.c_uint8), ("len", C.c_uint8)]


class BlockSet(C.Structure):
    _fields_ = [("data", C.POINTER(C.c_uint8)),
                ("block_offsets", C.POINTER(C.c_uint64)),
                ("n_blocks", C.c_uint32), ("n_cols", C.c_uint16),
                ("cols", C.POINTER(ColSchema)), ("total_rows", C.c_uint64)]


class FilterLeaf(C.Structure):
    _fields_ = [("col", C.c_uint16), ("op", C.c_uint8), ("n_in", C.c_uint8),
                ("lo", C.c_int64), ("hi", C.c_int64),
                ("in_list", C.c_int64 * 8),
                # black (generic-expression) leaves, op == OP_BLACK
                ("bconst", C.c_int64 * 4), ("bcols", C.c_uint16 * 4),
                ("n_bprog", C.c_uint8), ("n_bcols", C.c_uint8),
                ("bprog", C.c_uint8 * 24), ("bpad", C.c_uint8 * 6)]


# black-filter bytecode (include/obx.h OBX_BX_*): postfix program over
# column values and constants; three-valued logic, wrap-mod-2^64 arith
BX_COL = 0x00   # | slot index (into the leaf's bcols)
BX_CONST = 0x40  # | const index
BX_ADD, BX_SUB, BX_MUL, BX_DIV, BX_NEG = 0x50, 0x51, 0x52, 0x53, 0x54
BX_MOD = 0x55
BX_LT, BX_LE, BX_GT, BX_GE, BX_EQ, BX_NE = 0x60, 0x61, 0x62, 0x63, 0x64, 0x65
BX_AND, BX_OR, BX_NOT = 0x70, 0x71, 0x72


TOK_AND, TOK_OR = 128, 129


class FilterDesc(C.Structure):
    _fields_ = [("n_leaves", C.c_uint16), ("n_prog", C.c_uint8),
                ("prog", C.c_uint8 * 15), ("leaves", FilterLeaf * 8)]


class AggExpr(C.Structure):
    _fields_ = [("kind", C.c_uint8), ("col_a", C.c_uint16),
                ("col_b", C.c_uint16), ("col_c", C.c_uint16)]


class AggDesc(C.Structure):
    _fields_ = [("n_group_cols", C.c_uint8), ("group_cols", C.c_uint16 * 2),
                ("n_aggs", C.c_uint8), ("aggs", AggExpr * 8)]


class AggCell(C.Structure):
    _fields_ = [("limb", C.c_uint64 * 4)]

    def as_int(self):
        """256-bit two's-complement -> Python int (exact)."""
        v = 0
        for i in range(3, -1, -1):
            v = (v << 64) | self.limb[i]
        if v >= 1 << 255:
            v -= 1 << 256
        return v


class GroupRow(C.Structure):
    _fields_ = [("key", C.c_uint8 * MAX_KEY_BYTES), ("key_len", C.c_uint8),
                ("row_count", C.c_uint64), ("cells", AggCell * 8)]


class AggResult(C.Structure):
    _fields_ = [("n_groups", C.c_uint32), ("groups", GroupRow * MAX_GROUPS),
                ("rows_scanned", C.c_uint64), ("rows_passed", C.c_uint64)]


def make_filter(leaves, prog=None):
    """leaves: list of dicts {col, op, lo, hi?, in_list?}; prog: optional
    postfix combine program (ints: leaf index, TOK_AND, TOK_OR); None = AND
    of all leaves."""
    f = FilterDesc()
    f.n_leaves = len(leaves)
    if prog:
        f.n_prog = len(prog)
        for i, t in enumerate(prog):
            f.prog[i] = t
    for i, lf in enumerate(leaves):
        if lf.get("op") == OP_BLACK or "bprog" in lf:
            bcols = lf["bcols"]
            prog = lf["bprog"]
            consts = lf.get("bconst", [])
            f.leaves[i].op = OP_BLACK
            f.leaves[i].col = bcols[0]
            f.leaves[i].n_bcols = len(bcols)
            for j, c in enumerate(bcols):
                f.leaves[i].bcols[j] = c
            f.leaves[i].n_bprog = len(prog)
            for j, b in enumerate(prog):
                f.leaves[i].bprog[j] = b
            for j, k in enumerate(consts):
                f.leaves[i].bconst[j] = k
            continue
        f.leaves[i].col = lf["col"]
        f.leaves[i].op = lf["op"]
        f.leaves[i].lo = lf.get("lo", 0)
        f.leaves[i].hi = lf.get("hi", 0)
        il = lf.get("in_list", [])
        f.leaves[i].n_in = len(il)
        for j, v in enumerate(il):
            f.leaves[i].in_list[j] = v
    return f


def make_agg(group_cols, aggs):
    """aggs: list of dicts {kind, col_a?, col_b?, col_c?}."""
    a = AggDesc()
    a.n_group_cols = len(group_cols)
    for i, c in enumerate(group_cols):
        a.group_cols[i] = c
    a.n_aggs = len(aggs)
    for i, e in enumerate(aggs):
        a.aggs[i].kind = e["kind"]
        a.aggs[i].col_a = e.get("col_a", 0xFFFF)
        a.aggs[i].col_b = e.get("col_b", 0)
        a.aggs[i].col_c = e.get("col_c", 0)
    return a


def group_row_tuples(rows, n_aggs):
    """[GroupRow] -> list of (key_bytes, row_count, [cell ints])."""
    out = []
    for g in rows:
        out.append((bytes(g.key[: g.key_len]), g.row_count,
                    [g.cells[a].as_int() for a in range(n_aggs)]))
    return out


def result_rows(res, n_aggs):
    """AggResult -> list of (key_bytes, row_count, [cell ints])."""
    out = []
    for i in range(res.n_groups):
        g = res.groups[i]
        key = bytes(g.key[: g.key_len])
        out.append((key, g.row_count,
                    [g.cells[a].as_int() for a in range(n_aggs)]))
    return out
