/*
 * obx.h — C-ABI drop-in boundary for the MI355X-native OceanBase OLAP hot path
 * (microblock decode → pushdown filter → aggregate).
 *
 * This header declares the surface a replacement ObTableScanOp/ObHashGroupByVecOp
 * engine exports, restated as a C ABI (the reference is in-process C++; we are
 * out-of-process, so the ObOperator verbs become batch calls over many
 * microblocks). Each entry point cites the reference interface it replaces:
 *
 *  - obx_filter_blocks:    ObIMicroBlockReader::filter_pushdown_filter
 *                          (/root/reference/src/storage/blocksstable/ob_imicro_block_reader.h:588-606),
 *                          batched over microblocks; result bitmap semantics of
 *                          ObPushdownFilterExecutor::execute
 *                          (/root/reference/src/sql/engine/basic/ob_pushdown_filter.cpp:1559-1632).
 *  - obx_decode_blocks:    ObIMicroBlockReader::get_rows
 *                          (/root/reference/src/storage/blocksstable/ob_imicro_block_reader.h:546-586):
 *                          project row_ids -> fixed-length column vectors
 *                          (VEC_FIXED of /root/reference/src/share/vector/type_traits.h:16-25).
 *  - obx_scan_filter_agg:  the fused ObTableScanOp -> ObHashGroupByVecOp /
 *                          ObScalarAggregateVecOp pipeline
 *                          (/root/reference/src/sql/engine/table/ob_table_scan_op.cpp:3034,
 *                           /root/reference/src/sql/engine/aggregate/ob_hash_groupby_vec_op.cpp:1400,
 *                           /root/reference/src/share/aggregate/processor.h:47).
 *  - obx_filter_desc:      POD restatement of the serialized ObPushdownFilterNode
 *                          tree (/root/reference/src/sql/engine/basic/ob_pushdown_filter.h:590,1550),
 *                          white filters only (ob_pushdown_filter.h:383-397).
 *  - obx_batch:            mirrors ObBatchRows{skip_,size_,end_,all_rows_active_}
 *                          (/root/reference/src/sql/engine/ob_batch_rows.h:19-67).
 *
 * Error model: int return codes, 0 = OBX_SUCCESS (no exceptions), like
 * OB_SUCCESS / OB_ITER_END (/root/reference/deps/oblib/src/lib/ob_errno.h).
 * Threading: one obx_ctx == one HIP stream == one caller thread, parallelism by
 * context replication (the reference parallelizes by PX operator replication).
 * Ownership: caller owns host buffers; the engine owns device scratch.
 *
 * Two implementations export this ABI:
 *   liboracle.so  — CPU restatement of the reference engine (oracle/; also the
 *                   host-cores baseline timed by bench.py).
 *   libobx.so     — the MI355X product path (oceanbase_amd/csrc; HIP/gfx950).
 */
#ifndef OBX_H_
#define OBX_H_

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status codes ------------------------------------------------------- */
enum {
  OBX_SUCCESS = 0,
  OBX_ITER_END = -4008,          /* OB_ITER_END */
  OBX_INVALID_ARGUMENT = -4002,  /* OB_INVALID_ARGUMENT */
  OBX_NOT_SUPPORTED = -4007,
  OBX_BUF_NOT_ENOUGH = -4009,
  OBX_NO_GPU = -7001,            /* product path refused: HIP device required */
  OBX_INTERNAL_ERROR = -4016,
  OBX_PHYSIC_CHECKSUM_ERROR = -4108, /* OB_PHYSIC_CHECKSUM_ERROR */
};

/* ---- column encodings (ObColumnHeader::Type,
 *      /root/reference/src/storage/blocksstable/ob_block_sstable_struct.h:196-209) */
enum obx_encoding {
  OBX_ENC_RAW = 0,
  OBX_ENC_DICT = 1,
  OBX_ENC_RLE = 2,
  OBX_ENC_CONST = 3,
  OBX_ENC_INTEGER_BASE_DIFF = 4,
  OBX_ENC_STRING_DIFF = 5,       /* fixed char: common/diff byte runs
                                    (ob_string_diff_encoder.h) */
  OBX_ENC_HEX_PACKING = 6,       /* fixed char, <=16 distinct chars: nibble
                                    packing (ob_hex_string_encoder.h) */
  OBX_ENC_STRING_PREFIX = 7,    /* fixed char: prefix table + suffixes
                                    (ob_string_prefix_encoder.h) */
  OBX_ENC_COLUMN_EQUAL = 8,
  OBX_ENC_COLUMN_SUBSTR = 9,     /* substring of a previous char column
                                    (ob_inter_column_substring_encoder.h) */      /* equals the nearest previous same-shape
                                    column except listed exception rows
                                    (ob_column_equal_encoder.h) */
  OBX_ENC_MAX = 10,
  OBX_ENC_AUTO = 255,            /* writer picks (cost-ranked like
                                    ob_encoding_util.h:270-303) */
};

/* ---- object types (subset of ObObjType,
 *      /root/reference/deps/oblib/src/common/object/ob_obj_type.h) */
enum obx_obj_type {
  OBX_T_INT = 5,          /* bigint, 8 B   (ObIntType)        */
  OBX_T_INT32 = 4,        /* int32, 4 B    (ObInt32Type)      */
  OBX_T_DATE = 19,        /* date, 4 B     (ObDateType)       */
  OBX_T_CHAR = 23,        /* char(N)       (ObCharType)       */
  OBX_T_DECIMAL_INT = 50, /* decimal-int   (ObDecimalIntType) */
};

/* ---- white filter ops (ObWhiteFilterOperatorType,
 *      /root/reference/src/sql/engine/basic/ob_pushdown_filter.h:383-397) */
enum obx_white_op {
  OBX_OP_EQ = 0,
  OBX_OP_LE = 1,
  OBX_OP_LT = 2,
  OBX_OP_GE = 3,
  OBX_OP_GT = 4,
  OBX_OP_NE = 5,
  OBX_OP_BT = 6,   /* between [lo, hi], both inclusive */
  OBX_OP_IN = 7,   /* in-list (operands in obx_filter_leaf.in_list) */
  OBX_OP_NU = 8,   /* is null */
  OBX_OP_NN = 9,   /* is not null */
  OBX_OP_BLACK = 10, /* black (generic-expression) filter: the leaf carries
                        a postfix program over column values and constants
                        (ObPhysicalFilterExecutor::filter_batch,
                        ob_pushdown_filter.cpp:2066 — decode the referenced
                        columns, evaluate the expression, keep is-true
                        rows). SQL three-valued logic: NULL operands
                        propagate; a NULL or false result drops the row. */
};

/* black-filter bytecode (one byte per op):
 *   0x00|i  push column value i (obx_filter_leaf.bcols[i])
 *   0x40|i  push constant i     (obx_filter_leaf.bconst[i])
 *   0x50 ADD  0x51 SUB  0x52 MUL  0x53 DIV (int; x/0 -> NULL)  0x54 NEG
 *   0x55 MOD (x%0 -> NULL; INT64_MIN%-1 == 0)
 *   0x60 LT  0x61 LE  0x62 GT  0x63 GE  0x64 EQ  0x65 NE (-> bool)
 *   0x70 AND  0x71 OR  0x72 NOT (three-valued)
 * Stack depth <= 8. The program leaves one value; the row passes iff it
 * is non-NULL and != 0. */
#define OBX_BX_MAX_PROG 24
#define OBX_BX_MAX_CONST 4
#define OBX_BX_MAX_COLS 4

/* ---- aggregate kinds (subset the hot path needs; AVG is rewritten to
 *      SUM+COUNT before the engine, ob_expand_aggregate_utils.cpp:1674) */
enum obx_agg_kind {
  OBX_AGG_COUNT = 0,      /* count(*) or count(col): int64 */
  OBX_AGG_SUM = 1,        /* sum(col): int128/int256 limbs (sum.h:49,72,213) */
  OBX_AGG_MIN = 2,
  OBX_AGG_MAX = 3,
  OBX_AGG_SUM_PROD2 = 4,  /* sum(a*(scale2_one - b)): Q1 disc_price */
  OBX_AGG_SUM_PROD3 = 5,  /* sum(a*(one-b)*(one+c)):  Q1 charge    */
  OBX_AGG_SUM_MUL = 6,    /* sum(a*b): Q6 revenue */
};

/* ---- schema ------------------------------------------------------------- */
typedef struct obx_col_schema {
  uint8_t obj_type;    /* enum obx_obj_type */
  int8_t scale;        /* decimal scale (datum int units = value*10^scale) */
  uint8_t precision;   /* decimal precision */
  uint8_t len;         /* fixed byte length of a datum of this column:
                          8 int/decimal64, 4 date/int32, N for char(N) */
} obx_col_schema;

/* ---- microblock set -----------------------------------------------------
 * Concatenated PAX microblocks in the byte format of
 * ObMicroBlockEncoder::build_block
 * (/root/reference/src/storage/blocksstable/encoding/ob_micro_block_encoder.cpp:492-554):
 * [ObMicroBlockHeader][ObColumnHeader xN][encoding meta + fix data].
 * block_offsets[i] is the byte offset of block i in `data`;
 * block_offsets[n_blocks] is the total byte size. */
typedef struct obx_blockset {
  const uint8_t *data;
  const uint64_t *block_offsets;  /* n_blocks+1 entries */
  uint32_t n_blocks;
  uint16_t n_cols;
  const obx_col_schema *cols;     /* n_cols entries */
  uint64_t total_rows;
} obx_blockset;

/* ---- filter descriptor (white-only AND tree for round 1) ---------------- */
#define OBX_MAX_IN_LIST 8
typedef struct obx_filter_leaf {
  uint16_t col;      /* column index (BLACK: bcols[0], for pruning) */
  uint8_t op;        /* enum obx_white_op */
  uint8_t n_in;      /* operand count for IN */
  int64_t lo;        /* operand (EQ/LT/...), or BT lower bound */
  int64_t hi;        /* BT upper bound */
  int64_t in_list[OBX_MAX_IN_LIST];
  /* BLACK leaves only (op == OBX_OP_BLACK) */
  int64_t bconst[OBX_BX_MAX_CONST];
  uint16_t bcols[OBX_BX_MAX_COLS];
  uint8_t n_bprog;
  uint8_t n_bcols;
  uint8_t bprog[OBX_BX_MAX_PROG];
  uint8_t bpad[6];
} obx_filter_leaf;

/* Filter combine program: the POD restatement of the executor tree's
 * AND/OR combine (ObPushdownFilterExecutor::execute,
 * ob_pushdown_filter.cpp:1559-1632). Postfix tokens over leaf bitmaps:
 *   0..n_leaves-1 : push leaf i's result
 *   OBX_TOK_AND   : pop two, push AND
 *   OBX_TOK_OR    : pop two, push OR
 * n_prog == 0 means AND of all leaves (the common conjunctive shape). */
#define OBX_TOK_AND 128
#define OBX_TOK_OR 129
#define OBX_MAX_PROG 15

typedef struct obx_filter_desc {
  uint16_t n_leaves;              /* 0 = no filter (all rows pass) */
  uint8_t n_prog;                 /* postfix program length (0 = AND-all) */
  uint8_t prog[OBX_MAX_PROG];
  obx_filter_leaf leaves[8];
} obx_filter_desc;

/* ---- aggregate descriptor ----------------------------------------------- */
typedef struct obx_agg_expr {
  uint8_t kind;     /* enum obx_agg_kind */
  uint16_t col_a;   /* input column (ignored for COUNT(*): col_a = UINT16_MAX) */
  uint16_t col_b;   /* second input for SUM_PROD2/3, SUM_MUL */
  uint16_t col_c;   /* third input for SUM_PROD3 */
} obx_agg_expr;

typedef struct obx_agg_desc {
  uint8_t n_group_cols;           /* 0 = scalar aggregate (Q6) */
  uint16_t group_cols[2];
  uint8_t n_aggs;
  obx_agg_expr aggs[8];
} obx_agg_desc;

/* ---- outputs ------------------------------------------------------------ */
/* One aggregate cell: 256-bit little-endian limbs (covers int64/128/256 sums;
 * mirrors the flat AggrRow cells of
 * /root/reference/src/share/aggregate/agg_ctx.h:21,237-253). */
typedef struct obx_agg_cell { uint64_t limb[4]; } obx_agg_cell;

#define OBX_MAX_GROUPS 64
#define OBX_MAX_KEY_BYTES 16
typedef struct obx_group_row {
  uint8_t key[OBX_MAX_KEY_BYTES]; /* concatenated group key datums */
  uint8_t key_len;
  uint64_t row_count;             /* rows aggregated into this group */
  obx_agg_cell cells[8];          /* one per obx_agg_desc.aggs entry */
} obx_group_row;

typedef struct obx_agg_result {
  uint32_t n_groups;
  obx_group_row groups[OBX_MAX_GROUPS];
  uint64_t rows_scanned;
  uint64_t rows_passed;
} obx_agg_result;

/* ---- batch descriptor (ObBatchRows mirror, ob_batch_rows.h:19-67) ------- */
typedef struct obx_batch {
  uint64_t *skip_bits;   /* 1 bit per row, 1 = SKIP (ObBitVector polarity) */
  int64_t size;
  uint8_t end;
  uint8_t all_rows_active;
} obx_batch;

/* ========================================================================= */
/* CPU oracle + reference-restatement engine (liboracle.so)                  */
/* ========================================================================= */

/* --- microblock writer (ObMicroBlockEncoder::build_block restatement) ---- */
/* Encode `row_count` rows of `n_cols` columns into one microblock at `out`.
 * col_data[c] points to row_count datums, each cols[c].len bytes, contiguous.
 * null_bitmaps[c] is NULL (no nulls) or row_count bits (1 = NULL).
 * enc_request[c] is an obx_encoding or OBX_ENC_AUTO.
 * Returns bytes written, or a negative status. */
int64_t obx_encode_block(const obx_col_schema *cols, uint16_t n_cols,
                         const uint8_t *const *col_data,
                         const uint8_t *const *null_bitmaps,
                         uint32_t row_count, const uint8_t *enc_request,
                         uint8_t *out, int64_t out_cap);

/* Payload checksum restating ob_crc64_sse42 semantics (CRC-32C in a u64
 * accumulator — deps/oblib/src/lib/checksum/ob_crc64.cpp:448): the value of
 * ObMicroBlockHeader::data_checksum_ over the bytes after the micro header.
 * Exported by BOTH libraries (independent implementations). */
uint64_t obx_crc32c(const uint8_t *buf, int64_t len);

/* --- microblock reader (ObMicroBlockDecoder restatement) ----------------- */
/* Decode whole block: for each requested column, write row_count datums of
 * cols[c].len bytes into out_cols[c] and (if out_nulls[c] != NULL) the null
 * bitmap. Returns OBX_SUCCESS. */
int obx_decode_block(const obx_col_schema *cols, uint16_t n_cols,
                     const uint8_t *block, int64_t block_len,
                     const uint16_t *proj_cols, uint16_t n_proj,
                     uint8_t *const *out_cols, uint8_t *const *out_nulls,
                     uint32_t *row_count);

/* --- pushdown filter over one block -------------------------------------- */
/* result_bits: 1 bit per row, 1 = row passes (ObBitmap polarity of
 * filter_pushdown_filter). Caller allocates ceil(rows/8)+8 bytes. */
int obx_cpu_filter_block(const obx_col_schema *cols, uint16_t n_cols,
                         const uint8_t *block, int64_t block_len,
                         const obx_filter_desc *filter,
                         uint8_t *result_bits, uint32_t *row_count,
                         uint32_t *popcnt);

/* --- fused scan->filter->aggregate over a blockset ----------------------- */
/* nthreads <= 0: use all host cores (per-core microblock work stealing). */
int obx_cpu_scan_filter_agg(const obx_blockset *bs,
                            const obx_filter_desc *filter,
                            const obx_agg_desc *agg,
                            int nthreads,
                            obx_agg_result *out);

/* --- synthetic TPC-H lineitem generator ---------------------------------- */
/* Writes microblocks for `row_count` rows of the 7-column Q1 lineitem layout
 * (or the config-2/3 layouts, see obx_gen.c) into a malloc'd buffer.
 * Column set + encodings are chosen by `config` (2, 3, 4=Q1, 6=Q6 per
 * BASELINE.json configs). Caller frees *out_data / *out_offsets via free().
 * Returns n_blocks or negative status. */
int64_t obx_gen_lineitem(int config, uint64_t row_count, uint64_t seed,
                         uint32_t target_block_bytes,
                         uint64_t row_id_base,
                         uint8_t **out_data, uint64_t **out_offsets,
                         obx_col_schema *out_cols, uint16_t *out_n_cols);

/* ========================================================================= */
/* MI355X product engine (libobx.so) — same verbs, GPU-resident              */
/* ========================================================================= */
typedef struct obx_gpu_ctx obx_gpu_ctx;

int obx_gpu_open(int device, obx_gpu_ctx **ctx);
int obx_gpu_close(obx_gpu_ctx *ctx);

/* Stage a blockset into HBM (one large contiguous device buffer + device
 * offset/descriptor arrays). Returns a handle id >= 0 or negative status. */
int obx_gpu_load_blocks(obx_gpu_ctx *ctx, const obx_blockset *bs);

/* CS (cs_encoding) blockset: `data`/`block_offsets` hold CS-format micro
 * blocks. The load-time transform of ObCSMicroBlockTransformer
 * (cs_encoding/ob_cs_micro_block_transformer.cpp) runs GPU-native: the
 * host parses stream metadata, device kernels decode every stream (RAW
 * widths, the PFoR/RLE codec families, dict refs, null recovery) into an
 * HBM arena in the engine's scan layout. Same handle semantics as
 * obx_gpu_load_blocks; scans/filters/aggregates then run unchanged. */
int obx_gpu_load_cs_blocks(obx_gpu_ctx *ctx, const obx_blockset *bs);

/* host-side CS block header probe (rows/cols); also exported for tests */
int obx_cs_host_parse(const uint8_t *buf, int64_t len, uint32_t *rows_out,
                      uint32_t *ncols_out);
int obx_gpu_free_blocks(obx_gpu_ctx *ctx, int handle);

/* filter_pushdown_filter equivalent: result bitmap + per-block pass counts +
 * compacted row ids (selection vector of ObVectorStore::fill_output_rows /
 * get_row_ids, ob_block_batched_row_store.cpp:107). Outputs stay on device;
 * obx_gpu_fetch_* copies them back for parity tests. */
int obx_gpu_filter(obx_gpu_ctx *ctx, int handle, const obx_filter_desc *filter,
                   int want_row_ids);
int obx_gpu_fetch_bitmap(obx_gpu_ctx *ctx, int handle, uint8_t *out, int64_t cap);
int obx_gpu_fetch_row_ids(obx_gpu_ctx *ctx, int handle, int32_t *out,
                          int64_t cap, uint64_t *n_out);
int obx_gpu_fetch_blk_counts(obx_gpu_ctx *ctx, int handle, uint32_t *out,
                             int64_t cap);

/* get_rows equivalent: decode projected columns for all rows into device
 * VEC_FIXED arrays; fetch for parity. */
int obx_gpu_decode(obx_gpu_ctx *ctx, int handle, const uint16_t *proj_cols,
                   uint16_t n_proj);
int obx_gpu_fetch_col(obx_gpu_ctx *ctx, int handle, uint16_t col,
                      uint8_t *out, int64_t cap);

/* Fused scan->filter->aggregate (the benchmark path). */
int obx_gpu_scan_filter_agg(obx_gpu_ctx *ctx, int handle,
                            const obx_filter_desc *filter,
                            const obx_agg_desc *agg,
                            obx_agg_result *out);

/* Timing helpers: milliseconds of device time of the last
 * filter/scan_filter_agg call (HIP events on the context stream). */
double obx_gpu_last_kernel_ms(obx_gpu_ctx *ctx);
/* device time of the last query's prep phase (plan upload +
 * k_lower_leaves) — with last_kernel_ms, the per-operator stats the
 * reference surfaces through sql_plan_monitor. */
double obx_gpu_last_prep_ms(obx_gpu_ctx *ctx);
/* 1 if the last scan_filter_agg ran the hipRTC plan-specialized kernel
 * (query codegen), 0 if the precompiled generic kernels ran. */
int obx_gpu_last_jit(obx_gpu_ctx *ctx);
uint64_t obx_gpu_total_rows(obx_gpu_ctx *ctx, int handle);
uint64_t obx_gpu_total_bytes(obx_gpu_ctx *ctx, int handle);
uint64_t obx_gpu_last_survivors(obx_gpu_ctx *ctx, int handle);

/* Paged access to the last scan's full sorted group rows — the growth
 * path past the OBX_MAX_GROUPS inline result. The reference's hash
 * group-by grows its table unboundedly
 * (ob_exec_hash_struct_vec.h:1718); here a scan whose group count
 * exceeds OBX_MAX_GROUPS returns OBX_BUF_NOT_ENOUGH with
 * obx_agg_result.n_groups set to the true total, and the rows (sorted
 * by key bytes) are fetched in pages. Device capacity: OBX_GTABLE_BIG
 * (4096) distinct groups; the engine transparently reruns an
 * LDS-table-overflowed generic scan through a direct-global
 * high-cardinality kernel (AND-combined filters only). */
int obx_gpu_agg_fetch(obx_gpu_ctx *ctx, int handle, uint32_t start,
                      uint32_t count, obx_group_row *out, uint32_t *n_out,
                      uint64_t *n_total);
/* CPU-oracle mirror (test infrastructure) */
int obx_cpu_agg_fetch(uint32_t start, uint32_t count, obx_group_row *out,
                      uint32_t *n_out, uint64_t *n_total);

/* TEST INFRASTRUCTURE: render the hipRTC source the JIT would generate
 * for a plan over a synthetic one-block handle (no GPU needed — codegen
 * and strategy selection are host code). agg == NULL renders the
 * bitmap-filter kernel. Returns source bytes (0 = plan not
 * JIT-eligible), negative on invalid arguments. */
int64_t obx_jit_dump_src(const obx_filter_desc *filter,
                         const obx_agg_desc *agg,
                         const obx_col_schema *cols, uint16_t n_cols,
                         const uint8_t *col_flags, const int64_t *col_min,
                         const int64_t *col_max, const uint32_t *col_maxcnt,
                         const uint32_t *col_maxw, uint32_t max_block_rows,
                         int force_v1, char *out, int64_t cap);

#ifdef __cplusplus
}
#endif
#endif /* OBX_H_ */
