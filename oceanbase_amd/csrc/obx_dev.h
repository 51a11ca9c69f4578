/*
 * obx_dev.h — device-side descriptors shared by the host engine
 * (obx_engine.cpp) and the gfx950 kernels (obx_kernels.hip).
 *
 * The host parses each microblock's headers ONCE at load time (the
 * reference's ObMicroBlockDecoder caches per-column decoders the same way,
 * ob_micro_block_decoder.cpp:440-480 cached_decoder_) and translates the
 * white-filter tree into per-block packed-domain tests:
 *   - dict/RLE columns: filter evaluated once over <=64 dict entries ->
 *     64-bit ref mask (the reference's dict_cmp_ref_func / black-filter-on-
 *     dict design, ob_dict_decoder.cpp:810-886,1481-1561)
 *   - bit-packed RAW / INTEGER_BASE_DIFF: compare mapped to an inclusive
 *     range (+invert) in the packed (unsigned) domain
 *   - fixed RAW: generic sign-extended int64 compare in-kernel
 * so the kernels do only coalesced reads + bit unpack + mask/range tests.
 */
#ifndef OBX_DEV_H_
#define OBX_DEV_H_

#if defined(__HIPRTC__) || defined(OBX_HIPRTC)
/* hipRTC has no libc headers; these match the HIP device ABI */
typedef signed char int8_t;
typedef unsigned char uint8_t;
typedef short int16_t;
typedef unsigned short uint16_t;
typedef int int32_t;
typedef unsigned int uint32_t;
typedef long long int64_t;
typedef unsigned long long uint64_t;
typedef unsigned long size_t;
#define INT64_MAX 0x7fffffffffffffffll
#define INT64_MIN (-INT64_MAX - 1ll)
#define UINT64_MAX 0xffffffffffffffffull
#else
#include <stdint.h>
#endif

#define OBX_DEV_MAX_COLS 8
#define OBX_DEV_MAX_LEAVES 8
#define OBX_DEV_MAX_AGGS 8
#define OBX_GTABLE_SLOTS 256   /* global group table (open addressing);
                                  LDS-mirrored by the pass kernels */
#define OBX_GTABLE_SHIFT 8     /* log2(OBX_GTABLE_SLOTS) */
#define OBX_GTABLE_BIG 4096    /* the high-cardinality direct kernel's
                                  table: growth past the 64-group inline
                                  result (ob_exec_hash_struct_vec.h:1718
                                  grows unboundedly — callers page with
                                  obx_gpu_agg_fetch) */
#define OBX_GTABLE_BIG_SHIFT 12
#define OBX_LTABLE_SLOTS 16    /* per-workgroup LDS group table */

/* column encodings (== ObColumnHeader::Type subset) */
enum {
  OBX_D_RAW = 0,
  OBX_D_DICT = 1,
  OBX_D_RLE = 2,
  OBX_D_CONST = 3,
  OBX_D_INTDIFF = 4,
  OBX_D_SDIFF = 5,   /* STRING_DIFF: common/diff byte runs */
  OBX_D_HEX = 6,     /* HEX_PACKING: nibble-packed chars */
  OBX_D_SPREFIX = 7, /* STRING_PREFIX: prefix table + suffixes */
  OBX_D_EQUAL = 8,   /* COLUMN_EQUAL: ref column + exception rows */
  OBX_D_SUBSTR = 9,  /* COLUMN_SUBSTR: slice of ref column + exceptions */
};

/* dev_col flags */
enum {
  OBX_DF_BITPACK = 1,   /* values/refs/diffs are a bit stream */
  OBX_DF_HAS_EXT = 2,   /* RAW/INTDIFF: ext (null) bits precede values */
  OBX_DF_SIGNED = 4,    /* sign-extend fixed values from store size */
  OBX_DF_STRING = 8,    /* char column: zero-extended bytes, binary order */
};

/* Per (block, column) descriptor. All *_bit offsets are absolute BIT
 * positions in the staged device buffer (buffer base is 256-B aligned, so
 * aligned-u64 + funnel-shift reads are always legal). Byte offsets are
 * absolute too. */
typedef struct dev_col {
  uint64_t data_bit;    /* RAW/INTDIFF: value bit stream (after ext bits);
                           DICT: ref bit stream */
  uint64_t ext_bit;     /* ext-bit stream start (valid if HAS_EXT) */
  uint64_t dict_byte;   /* dict payload (DICT/RLE/CONST-with-exceptions) */
  uint64_t aux_byte;    /* RLE: run row_ids start; CONST: exc refs start
                           (exc row_ids at aux_byte + runs * rib) */
  int64_t base;         /* INTDIFF base (sign-extended); CONST: the decoded
                           const value (int64) for the no-exception path */
  uint32_t count;       /* dict count; CONST: 0 = null-const block */
  uint32_t runs;        /* RLE run count; CONST: exception count */
  uint8_t enc;
  uint8_t flags;
  uint8_t width;        /* value/ref/diff width: bits if BITPACK else bytes */
  uint8_t entry_len;    /* dict entry byte len */
  uint8_t rib, rfb;     /* RLE row-id/ref byte widths; CONST: rib, const_ref */
  uint8_t datum_len;    /* output datum byte len (schema len) */
  uint8_t ext_width;    /* block extend_value_bit (1 or 2) when HAS_EXT */
  uint8_t tss;          /* type store size for sign-extension (4 or 8) */
  uint8_t pad[7];
} dev_col;

typedef struct dev_block {
  uint32_t row_start_lo;   /* global row index of row 0 of this block */
  uint32_t row_start_hi;
  uint32_t row_count;
  uint32_t block_len;      /* block byte length (for LDS staging) */
  uint64_t block_byte;     /* block start offset in the staged buffer
                              (16-B aligned when the LDS kernels are used) */
  dev_col cols[OBX_DEV_MAX_COLS];
} dev_block;

/* max block size eligible for LDS staging (kernel stages the whole block) */
#define OBX_LDS_STAGE_BYTES 17408

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
__host__ __device__
#endif
static inline uint64_t dev_block_row_start(const dev_block *b) {
  return ((uint64_t)b->row_start_hi << 32) | b->row_start_lo;
}

/* per-(block, leaf) precomputed test mode */
enum {
  OBX_LEAF_NONE = 0,     /* no row can pass */
  OBX_LEAF_ALL = 1,      /* every (non-null) row passes */
  OBX_LEAF_REF_MASK = 2, /* dict/RLE: bit r of mask = ref r passes
                            (null ref >= count handled by mask bit absent) */
  OBX_LEAF_RANGE = 3,    /* packed-domain unsigned range [lo,hi], ^invert */
  OBX_LEAF_VALUE = 4,    /* generic in-kernel int64 compare */
  OBX_LEAF_NULL = 5,     /* NU/NN on ext-bit columns: pass = is_null^invert */
};

/* global part of a leaf (one per filter leaf, shared by every block) */
typedef struct dev_leaf {
  int64_t vlo, vhi;     /* VALUE-mode operands (order-mapped for char) */
  uint16_t col;
  uint8_t op;           /* obx_white_op for VALUE mode (10 = BLACK) */
  uint8_t n_in;
  uint8_t char_len;     /* >0: char column — order-map values before compare
                           (byte-lexicographic, like the oracle's char_key) */
  uint8_t pad[3];
  int64_t in_list[8];   /* VALUE-mode IN */
  /* BLACK (op 10): postfix program over bcols values + bconst
     (ob_pushdown_filter.cpp:2066 semantics; obx.h OBX_BX_* bytecode) */
  int64_t bconst[4];
  uint16_t bcols[4];
  uint8_t n_bprog;
  uint8_t n_bcols;
  uint8_t bprog[24];
  uint8_t bpad[6];
} dev_leaf;

/* per-(block, leaf) lowered test (32 B, n_blocks * n_leaves of them) */
typedef struct blk_leaf {
  uint64_t mask;        /* REF_MASK */
  uint64_t lo, hi;      /* RANGE (packed domain, unsigned) */
  uint8_t mode;
  uint8_t invert;
  uint8_t pad[6];
} blk_leaf;

/* aggregate expr: inputs are indices into the per-row decoded-value array
 * (val_idx space = plan need_cols) */
typedef struct dev_agg {
  uint8_t kind;         /* obx_agg_kind */
  uint8_t ia, ib, ic;   /* indices into decoded vals (0xFF = none) */
  int32_t pad;
  int64_t one_b, one_c; /* 10^scale constants for PROD2/PROD3 */
} dev_agg;

#define OBX_DEV_MAX_NEED 12

/* aggregate-pass grouping: aggs whose inputs share <=3 distinct columns run
 * in one row pass (engine builds these in prep_query) */
typedef struct dev_pass_agg {
  uint8_t kind;        /* obx_agg_kind */
  uint8_t agg_idx;     /* output cell index */
  uint8_t sa, sb, sc;  /* selectors into the pass's decoded values (0..2) */
  uint8_t pad[3];
  int64_t one_b, one_c;
} dev_pass_agg;

typedef struct dev_pass {
  uint8_t n_aggs;
  uint8_t n_cols;      /* distinct decoded columns (<=3) */
  uint8_t cols[3];     /* need-slot index per decoded value */
  uint8_t pad[3];
  dev_pass_agg aggs[OBX_DEV_MAX_AGGS];
} dev_pass;

/* plan header, passed to kernels by value */
typedef struct dev_plan_hdr {
  uint32_t n_leaves;
  uint32_t n_aggs;
  uint32_t n_group_cols;
  uint32_t n_need;              /* decoded columns per surviving row */
  uint32_t n_passes;
  uint8_t n_prog;               /* filter combine program (0 = AND-all) */
  uint8_t prog[15];             /* postfix: leaf idx | 128=AND | 129=OR */
  uint16_t need_cols[OBX_DEV_MAX_NEED]; /* column index per val slot */
  uint8_t group_idx[2];         /* val-slot index of group cols */
  uint8_t group_len[2];         /* datum byte lens of group cols */
  dev_agg aggs[OBX_DEV_MAX_AGGS];
  dev_pass passes[OBX_DEV_MAX_AGGS];
} dev_plan_hdr;

/* global group table slot: key + count + n_aggs 256-bit cells.
 * key 0xFFFF.. = empty (keys are <=8 packed datum bytes; all-ones key is
 * not produced by <=8-byte keys because key_len < 8 zero-pads). */
typedef struct gslot {
  unsigned long long key;
  unsigned long long count;
  unsigned long long cells[OBX_DEV_MAX_AGGS][4];
} gslot;

#define OBX_KEY_EMPTY 0xFFFFFFFFFFFFFFFFull

#endif /* OBX_DEV_H_ */
