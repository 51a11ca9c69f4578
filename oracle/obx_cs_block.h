/*
 * obx_cs_block.h — CS (cs_encoding) COLUMN/BLOCK layer restatement,
 * continuing SURVEY §8(f) row 2 above the stream codecs in obx_cs.h.
 *
 * TEST INFRASTRUCTURE (oracle). Restates, with reference citations:
 *   ob_column_encoding_struct.h:24-132 — ObCSColumnHeader: 4 packed bytes
 *     {version, type, attrs, obj_type}; type INTEGER=0 / STRING=1; attrs
 *     IS_FIXED_LENGTH / HAS_NULL_OR_NOP_BITMAP / HAS_NOP(_BITMAP)
 *   ob_column_encoding_struct.h:134-169 — ObAllColumnHeader: 12 packed
 *     bytes {version, attrs, all_string_data_length, stream_offsets_length,
 *     stream_count}
 *   ob_micro_block_cs_encoder.cpp:1387-1480 (build_block) — block layout:
 *     [micro header][ObAllColumnHeader][ObCSColumnHeader x n]
 *     [per-column data][pooled all-string data][stream-offset stream]
 *   ob_icolumn_cs_encoder.cpp:93-141 — null/nop bitmaps are MSB-first
 *     (bit 1<<(7-r%8)) and precede the column's first stream
 *   ob_integer_column_encoder.cpp:100-285 — the null-replace decision:
 *     prefer a value ADJACENT to the data range (min-1, or max+1 when
 *     min==0) so decoders recover nulls by equality; fall back to a
 *     bitmap only when the range already spans the full store type
 *   ob_string_column_encoder.cpp:53-135 — fixed-vs-var decision: fixed
 *     columns with nulls choose bitmap+padding vs zero-len-as-null by
 *     estimated cost; var columns with no zero-length datum use
 *     zero-len-as-null; a zero-length real datum forces the bitmap
 *   ob_string_stream_encoder.{h,cpp} — string column stores only
 *     [string meta][offset int stream] in the column region; the bytes
 *     go to the pooled all-string region; offsets are cumulative END
 *     offsets, byte width = get_byte_packed_int_size(uncompressed_len)
 *   ob_micro_block_cs_encoder.cpp:1305-1366 (store_stream_offsets_) —
 *     the block tail holds every stream's absolute end offset as one
 *     more integer stream (monotonic, no base, width from the last one)
 *
 * Deviations (documented, deliberate): the leading micro header is the
 * obx 16-byte form below, not ObMicroBlockHeader (that header is generic
 * PAX/CS infrastructure with checksum/version fields outside this path);
 * the pooled string region is never compressed (NONE_COMPRESSOR) and the
 * block-tail offset stream is always RAW (the reference picks codecs by
 * ObCSEncodingOpt cost trials — codec CHOICE is policy, the formats are
 * what this slice pins; dict ref streams likewise always RAW).
 * INT_DICT/STR_DICT (ob_dict_column_encoder.cpp and subclasses) are
 * restated including the CONST_ENCODING_REF exception-list layout and
 * the ref width rule (width covers ref_stream_max_value_, not just the
 * stored array's max). SEMISTRUCT remains out of scope.
 */
#ifndef OBX_CS_BLOCK_H_
#define OBX_CS_BLOCK_H_

#include "obx_cs.h"

#ifdef __cplusplus
extern "C" {
#endif

#define OBX_CS_BLOCK_MAGIC 0x5343424Fu /* 'OBCS' little-endian */
#define OBX_CS_MAX_COLS 48
#define OBX_CS_MAX_STREAMS 96

/* ObCSColumnHeader::Type */
enum {
  OBX_CS_COL_INTEGER = 0,
  OBX_CS_COL_STRING = 1,
  OBX_CS_COL_INT_DICT = 2,
  OBX_CS_COL_STR_DICT = 3,
};

/* ObDictEncodingMeta::Attribute (ob_column_encoding_struct.h:171-208) */
enum {
  OBX_CS_DICT_IS_SORTED = 0x1,
  OBX_CS_DICT_HAS_NULL = 0x2,
  OBX_CS_DICT_CONST_REF = 0x4,
};

/* ObDictColumnEncoder limits (ob_dict_column_encoder.h:21-22) */
#define OBX_CS_DICT_MAX_EXCEPTIONS 64
#define OBX_CS_DICT_MAX_EXCEPTION_PCT 10

/* ObCSColumnHeader::Attribute */
enum {
  OBX_CS_CA_IS_FIXED = 0x01,
  OBX_CS_CA_HAS_NULL_BITMAP = 0x02,
};

/* ObObjType codes stored in the header's obj_type byte
 * (deps/oblib/src/common/object/ob_obj_type.h: ObIntType=5,
 * ObVarcharType=22) */
enum {
  OBX_OBJ_INT = 5,
  OBX_OBJ_VARCHAR = 22,
};

#pragma pack(push, 1)
typedef struct obx_cs_block_header {
  uint32_t magic;
  uint16_t version;
  uint16_t header_size;
  uint32_t row_count;
  uint16_t column_count;
  uint16_t reserved;
} obx_cs_block_header;

/* ObAllColumnHeader (ob_column_encoding_struct.h:134-169) */
typedef struct obx_cs_all_col_header {
  uint8_t version;
  uint8_t attrs;
  uint32_t all_string_data_length;
  uint32_t stream_offsets_length;
  uint16_t stream_count;
} obx_cs_all_col_header;

/* ObCSColumnHeader (ob_column_encoding_struct.h:60-63) */
typedef struct obx_cs_col_header {
  uint8_t version;
  uint8_t type;
  uint8_t attrs;
  uint8_t obj_type;
} obx_cs_col_header;

/* ObDictEncodingMeta (ob_column_encoding_struct.h:202-205): 10 packed
 * bytes ahead of the dict-value stream(s) */
typedef struct obx_cs_dict_meta {
  uint8_t version;
  uint8_t attrs;
  uint32_t distinct_val_cnt;
  uint32_t ref_row_cnt;
} obx_cs_dict_meta;
#pragma pack(pop)

/* encoder input: one column. Integer columns give ivals; string columns
 * give bytes (non-null rows' data concatenated in row order) + lens
 * (per-row, null rows ignored). nulls is LSB-first (obx convention; the
 * stored bitmap is MSB-first per the reference). enc_type 0 -> RAW. */
typedef struct obx_cs_col_in {
  uint8_t is_string;
  uint8_t enc_type;
  uint8_t want_dict; /* 1 -> INT_DICT/STR_DICT (encoding CHOICE is the
                        reference's cost-trial policy, encoder_detection_;
                        here the caller decides, the FORMAT is what is
                        restated) */
  uint8_t store_width; /* integer column store-type size in bytes
                          (1/2/4/8, 0 -> 8): the null-replace decision
                          compares min/max against THIS type's bounds
                          (ob_integer_column_encoder.cpp:183-186
                          INTEGER_MASK_TABLE[type_store_size_]), so an
                          int32 column whose min is INT32_MIN picks
                          max+1, not min-1 */
  const int64_t *ivals;
  const uint8_t *bytes;
  const uint32_t *lens;
  const uint8_t *nulls;
} obx_cs_col_in;

int64_t obx_cs_block_enc(uint32_t rows, uint32_t ncols,
                         const obx_cs_col_in *cols, uint8_t *buf,
                         size_t cap);

typedef struct obx_cs_col_view {
  obx_cs_col_header h;
  const uint8_t *null_bitmap; /* MSB-first in-block bitmap or NULL */
  const uint8_t *int_stream;  /* integer col / dict-value int stream */
  size_t int_stream_len;
  obx_cs_str_meta sm;         /* string col / str-dict value stream */
  const uint8_t *off_stream;  /* var string: offset int stream */
  size_t off_stream_len;
  uint32_t str_data_off;      /* into the pooled all-string region */
  obx_cs_dict_meta dm;        /* dict cols */
  const uint8_t *ref_stream;  /* dict cols: ref int stream */
  size_t ref_stream_len;
} obx_cs_col_view;

typedef struct obx_cs_block_view {
  const uint8_t *buf;
  size_t len;
  uint32_t rows;
  uint32_t ncols;
  obx_cs_all_col_header ach;
  const uint8_t *all_string;
  uint32_t stream_offsets[OBX_CS_MAX_STREAMS];
  uint32_t stream_count;
  obx_cs_col_view col[OBX_CS_MAX_COLS];
} obx_cs_block_view;

/* parse headers + stream boundaries; returns 0 or -1 */
int obx_cs_block_dec(const uint8_t *buf, size_t len, obx_cs_block_view *v);

/* materialize integer column c: out[rows]; nulls_out (LSB-first, may be
 * NULL if the caller knows the column has no nulls). Null rows decode
 * to 0. Returns 0 or -1. */
int obx_cs_block_get_int(const obx_cs_block_view *v, uint32_t c,
                         int64_t *out, uint8_t *nulls_out);

/* materialize string column c: row bytes concatenated into bytes_out
 * (cap bytes_cap), per-row lens into lens_out, nulls LSB-first into
 * nulls_out (may be NULL). Null rows have len fixed_len (zero-filled)
 * on fixed columns and 0 on var columns. Returns total bytes or -1. */
int64_t obx_cs_block_get_str(const obx_cs_block_view *v, uint32_t c,
                             uint8_t *bytes_out, size_t bytes_cap,
                             uint32_t *lens_out, uint8_t *nulls_out);

#ifdef __cplusplus
}
#endif
#endif /* OBX_CS_BLOCK_H_ */
