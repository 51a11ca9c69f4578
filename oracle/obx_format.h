/*
 * obx_format.h — byte-level PAX microblock format, restated from the
 * reference (oceanbase/oceanbase @ /root/reference). TEST INFRASTRUCTURE +
 * FORMAT CONTRACT: this header is shared by the CPU oracle (oracle/) and the
 * GPU engine's host-side block parser; the structs must stay bit-identical to
 * the cited reference definitions.
 *
 * PARITY PINNING NOTE: the reference cannot be built in this container
 * (deps/3rd toolchain RPMs are network-fetched; no network) and ships NO
 * stored golden vectors for this path — its own tests are in-process
 * encode->decode round-trip + computed-bitmap assertions
 * (unittest/storage/blocksstable/encoding/test_column_decoder.h:65-185).
 * Cross-binary parity is therefore UNPINNED IN-CONTAINER; we pin by
 * (a) struct-level fidelity to the definitions cited below (byte inspection),
 * (b) re-creating the reference's round-trip/popcount test assertions, and
 * (c) an independent pure-Python/numpy model in tests/ (three implementations
 * deep, mirroring the reference's test strategy). See DESIGN.md.
 *
 * Cited layouts:
 *  ObMicroBlockHeader     /root/reference/src/storage/blocksstable/ob_micro_block_header.h:86-140
 *                         (fixed 64-byte serialized prefix; COLUMN_CHECKSUM_PTR_OFFSET = 64,
 *                          ob_micro_block_header.cpp:14)
 *  ObColumnHeader         /root/reference/src/storage/blocksstable/ob_block_sstable_struct.h:194-231
 *  ObDictMetaHeader       /root/reference/src/storage/blocksstable/encoding/ob_dict_encoder.h:24-52
 *  ObRLEMetaHeader        /root/reference/src/storage/blocksstable/encoding/ob_rle_encoder.h:20-41
 *  ObConstMetaHeader      /root/reference/src/storage/blocksstable/encoding/ob_const_encoder.h:21-43
 *  ObIntegerBaseDiffHeader /root/reference/src/storage/blocksstable/encoding/ob_integer_base_diff_encoder.h:19-30
 *  bit stream             /root/reference/src/storage/blocksstable/encoding/ob_bit_stream.h:27-120
 *                         (little-endian bit order: bit i lives in byte i/8 at weight 1<<(i%8))
 *  block layout           /root/reference/src/storage/blocksstable/encoding/ob_micro_block_encoder.cpp:492-554:
 *                         [header][col headers][per-col: meta then fix data][row var data][row index]
 *                         row index/var data absent when all columns are fix-stored
 *                         (row_index_byte_ = 0, ob_micro_block_encoder.cpp:617-619).
 *  column fix-data layout /root/reference/src/storage/blocksstable/encoding/ob_icolumn_encoder.h:119-268:
 *                         [extend bits (ext_bit*rows)][bit-packed values (k*rows)] packed together,
 *                         rounded up to whole bytes ONCE, then [fixed data rows*len].
 */
#ifndef OBX_FORMAT_H_
#define OBX_FORMAT_H_

#include <stdint.h>
#include <string.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- ObMicroBlockHeader (64-byte serialized prefix) --------------------- */
#pragma pack(push, 1)
typedef struct obx_micro_header {
  int16_t magic;            /* MICRO_BLOCK_HEADER_MAGIC = 1005
                               (ob_block_sstable_struct.h:41) */
  int16_t version;          /* MICRO_BLOCK_HEADER_VERSION_LATEST = 3 */
  uint32_t header_size;     /* = 64 (no column checksums) */
  int16_t header_checksum;
  uint16_t column_count;
  uint16_t rowkey_column_count;
  uint16_t flag16;          /* bitfields: has_column_checksum etc.; 0 here */
  uint32_t row_count;
  uint8_t row_store_type;   /* ENCODING_ROW_STORE = 1 (ob_store_format.h:27) */
  uint8_t opt;              /* pax: row_index_byte_:3 | extend_value_bit_:3 */
  uint16_t var_column_count;
  uint32_t row_data_offset; /* offset of row (var) data region */
  int32_t original_length;
  int64_t max_merged_trans_version;
  int32_t data_length;
  int32_t data_zlength;     /* == data_length: uncompressed */
  int64_t data_checksum;
  /* up to here: 56 bytes */
  uint8_t reserved[8];      /* serialized column_checksums_ pointer slot
                               (always nullptr, ob_micro_block_header.cpp:120);
                               pads prefix to COLUMN_CHECKSUM_PTR_OFFSET = 64 */
} obx_micro_header;

#define OBX_MICRO_HEADER_SIZE 64
#define OBX_MICRO_BLOCK_MAGIC 1005
#define OBX_MICRO_BLOCK_VERSION 3
#define OBX_ROW_STORE_ENCODING 1

static inline uint8_t obx_hdr_row_index_byte(const obx_micro_header *h) {
  return (uint8_t)(h->opt & 0x7);
}
static inline uint8_t obx_hdr_extend_value_bit(const obx_micro_header *h) {
  return (uint8_t)((h->opt >> 3) & 0x7);
}
static inline void obx_hdr_set_opt(obx_micro_header *h, uint8_t row_index_byte,
                                   uint8_t extend_value_bit) {
  h->opt = (uint8_t)((row_index_byte & 0x7) | ((extend_value_bit & 0x7) << 3));
}

/* ---- ObColumnHeader (16 bytes) ------------------------------------------ */
enum {
  OBX_COL_RAW = 0,
  OBX_COL_DICT = 1,
  OBX_COL_RLE = 2,
  OBX_COL_CONST = 3,
  OBX_COL_INTEGER_BASE_DIFF = 4,
  OBX_COL_STRING_DIFF = 5,
  OBX_COL_HEX_PACKING = 6,
  OBX_COL_STRING_PREFIX = 7,
  OBX_COL_EQUAL = 8,
  OBX_COL_SUBSTR = 9,
  /* 5..9: string encoders, out of scope this round */
};
enum {
  OBX_COL_ATTR_FIX_LENGTH = 0x1,
  OBX_COL_ATTR_HAS_EXTEND_VALUE = 0x2,
  OBX_COL_ATTR_BIT_PACKING = 0x4,
  OBX_COL_ATTR_LAST_VAR_FIELD = 0x8,
};
typedef struct obx_col_header {
  int8_t version;            /* OB_COLUMN_HEADER_V1 = 0 */
  int8_t type;               /* enum above */
  int8_t attr;
  uint8_t obj_type;          /* ObObjType, 1 byte */
  uint32_t extend_value_offset; /* var-column null bitmap offset; 0 here */
  uint32_t offset;           /* start of this column's meta (RAW: fix data)
                                relative to meta region start */
  uint32_t length;           /* meta length (RAW/INT_DIFF fix: per-row byte len
                                or bit width; see encoder notes) */
} obx_col_header;

/* ---- per-encoding metas -------------------------------------------------- */
enum { OBX_DICT_ATTR_FIX_LENGTH = 0x1, OBX_DICT_ATTR_IS_SORTED = 0x2 };
typedef struct obx_dict_meta {
  uint8_t version;       /* 0 */
  uint8_t row_ref_size;  /* ref bit width (BIT_PACKING) or byte width */
  uint32_t count;        /* dict entry count (null ref = count, nope = count+1) */
  uint16_t data_size;    /* fix dict: entry byte len; var dict: index_byte */
  uint8_t attr;
  /* payload: fix dict: count*data_size bytes;
     var dict: (count-1)*index_byte offsets then var data */
} obx_dict_meta;         /* 9 bytes packed */

typedef struct obx_rle_meta {
  uint8_t version;       /* 0 */
  uint8_t attr;          /* row_id_byte_:3 | ref_byte_:3 */
  uint32_t count;        /* run count */
  uint32_t offset;       /* offset of the dict meta from this header */
  /* payload: [run start row_ids: count*row_id_byte][run refs: count*ref_byte]
     then at +offset: obx_dict_meta + dict payload */
} obx_rle_meta;          /* 10 bytes packed */

typedef struct obx_const_meta {
  uint8_t version;       /* 0 */
  uint8_t count;         /* exception count */
  uint8_t const_ref;     /* dict ref of the const value (0 when no dict) */
  uint8_t attr;          /* row_id_byte_:3 */
  uint16_t offset;       /* offset of dict meta (or const datum) from header */
  /* exceptions path (count>0): payload = [exc refs: count*1][exc row_ids:
     count*row_id_byte], then at +offset dict meta.  No-exception path:
     payload at +offset = the const datum bytes (no dict). */
} obx_const_meta;        /* 6 bytes packed */

/* HEX_PACKING meta (ObHexStringHeader, ob_hex_string_encoder.h:128-139;
   the reference's offset_/length_ var-data bookkeeping is dropped — our
   boundary stores fixed char(N<=8) only, so string_size is the fixed N).
   Followed by char_cnt bytes of hex_char_array (build_index order: the
   distinct chars ASCENDING; nibble k decodes as hex_char_array[k],
   ob_hex_string_encoder.cpp:22-34). Row data (fix region, after the usual
   ext bits): ceil(N/2) bytes per row, HIGH nibble first
   (ObHexStringPacker::pack, pos even -> shift 4). */
typedef struct obx_hex_meta {
  uint8_t version;       /* 0 */
  uint8_t char_cnt;      /* distinct chars, <= 16 */
  uint16_t string_size;  /* N (== schema len) */
} obx_hex_meta;          /* 4 bytes packed */

/* STRING_DIFF meta (ObStringDiffHeader, ob_string_diff_encoder.h:20-97).
   diff_descs: 1 byte each, bit0 = diff_ (1: positions vary per row),
   bits1-7 = count_ (run length; reference bitfield diff_:1,count_:7).
   Followed by: hex_char_array (hex_char_cnt bytes; >0 means the per-row
   diff bytes are nibble-packed exactly like HEX_PACKING), then
   common_data (the bytes of all non-diff runs, in position order).
   Row data (fix region): diff bytes of the row in position order,
   hex-packed when hex_char_cnt > 0; stride = diff_len or ceil(diff_len/2). */
typedef struct obx_sdiff_meta {
  uint8_t version;       /* 0 */
  uint8_t hex_char_cnt;  /* 0 = plain diff bytes */
  uint16_t string_size;  /* N */
  uint8_t diff_desc_cnt;
} obx_sdiff_meta;        /* 5 bytes packed */

/* STRING_PREFIX meta (ObStringPrefixMetaHeader,
   ob_string_prefix_encoder.h:64-105): a table of <=15 prefixes; each row
   stores a prefix ref + its suffix. Restated for the fixed char(N<=8)
   boundary with FIXED-STRIDE cells (the reference's var-cell len_ field
   is derivable for fixed-length strings, so cells carry only the ref
   nibble; suffixes are nibble-packed like HEX_PACKING when the suffix
   chars fit 16 — the odd_ nibble is implicit for fixed lengths).
   Layout after the header: hex_char_array (hex_char_cnt B), prefix END
   offsets (count entries x pib B, cumulative — the reference stores
   count-1, we keep all ends), prefix bytes concatenated. Row data (fix
   region): stride = 1 + max_suffix (raw) or 1 + ceil(max_suffix/2)
   (hex); byte 0 low nibble = prefix ref. Our writer emits equal-length
   prefixes (writer policy); the decoder honors per-prefix lengths via
   the offset index. */
typedef struct obx_sprefix_meta {
  uint8_t version;       /* 0 */
  uint8_t count;         /* prefixes, 1..15 */
  uint16_t string_size;  /* N */
  uint8_t hex_char_cnt;  /* 0 = raw suffix bytes */
  uint8_t pib;           /* prefix end-offset byte width */
} obx_sprefix_meta;      /* 6 bytes packed */

/* COLUMN_EQUAL meta (ObColumnEqualMetaHeader,
   ob_column_equal_encoder.h:24-40): this column equals ref_col except at
   the exception rows. The reference stores exceptions through its
   ObBitMapMeta family; our container reuses the CONST-style exception
   list instead (ascending row_ids + fixed-size datums) — same
   information, simpler layout, documented divergence. Exception NULLs
   are a bitmap between the row_ids and the datums. */
typedef struct obx_coleq_meta {
  uint8_t version;       /* 0 */
  uint16_t ref_col;      /* schema index of the reference column */
  uint16_t exc_cnt;      /* exception rows */
  uint8_t rib;           /* row-id byte width */
  /* then: exc row_ids (rib B each, ascending); exc null bitmap
     ceil(exc_cnt/8) B; exc datums (schema len B each, 0 for nulls) */
} obx_coleq_meta;        /* 6 bytes packed */

/* COLUMN_SUBSTR meta (ObInterColSubStrMetaHeader,
   ob_inter_column_substring_encoder.h:18-55): this column is a substring
   of ref_col. Restated for the fixed boundary with is_same_start_pos and
   is_fix_length always set (start_pos and the schema length describe the
   slice), so rows carry no per-row start/len; exceptions use the same
   CONST-style list as our COLUMN_EQUAL (ascending row_ids + null bitmap
   + datums — the reference's EXCEPTION_START_POS/EXT_START_POS markers
   folded into the list). */
typedef struct obx_substr_meta {
  uint8_t version;       /* 0 */
  uint8_t attr;          /* bit0 same_start (always 1), bit1 fix_len (1) */
  uint16_t start_pos;
  uint16_t ref_col;
  uint16_t exc_cnt;
  uint8_t rib;
  uint8_t ref_len;       /* ref column datum bytes (decoder has no schema
                            array in scope for the nested decode) */
} obx_substr_meta;       /* 10 bytes packed */

typedef struct obx_intdiff_meta {
  uint8_t version;       /* 0 */
  uint8_t length;        /* diff bit width (BIT_PACKING) or byte width */
  /* followed by base value, type_store_size bytes; then (in fix-data region
     immediately after) the [ext bits][packed diffs] stream */
} obx_intdiff_meta;      /* 2 bytes packed */
#pragma pack(pop)

/* ---- bit stream ---------------------------------------------------------
 * ObBitStream::memory_safe_set / get (ob_bit_stream.h:88-120): value v of
 * `len` bits stored at bit position `pos`; byte = pos/8, little-endian bit
 * order within bytes; writes may touch up to 9 bytes (the encoder reserves
 * +8 bytes of zeroed slack, ob_icolumn_encoder.h:243). */
static inline void obx_bs_set(uint8_t *buf, int64_t pos, int64_t len,
                              uint64_t v) {
  int64_t off = pos >> 3;
  int64_t word_off = pos & 7;
  uint64_t cur;
  memcpy(&cur, buf + off, 8);
  cur |= (v << word_off);
  memcpy(buf + off, &cur, 8);
  if (word_off + len > 64) {
    buf[off + 8] |= (uint8_t)(((1u << word_off) - 1) &
                              (v >> (64 - word_off)));
  }
}

static inline uint64_t obx_bs_get(const uint8_t *buf, int64_t pos,
                                  int64_t len) {
  /* ObBitStream::get DEFAULT unpack (ob_bit_stream.h:56-80): read 2 words */
  int64_t off = pos >> 3;
  int64_t word_off = pos & 7;
  uint64_t lo;
  memcpy(&lo, buf + off, 8);
  uint64_t v = lo >> word_off;
  if (word_off + len > 64) {
    uint64_t hi = buf[off + 8];
    v |= hi << (64 - word_off);
  }
  if (len < 64) v &= (((uint64_t)1 << len) - 1);
  return v;
}

/* INTEGER_MASK_TABLE (ob_encoding_util.cpp:25-28) */
static inline uint64_t obx_integer_mask(int64_t byte_len) {
  return byte_len >= 8 ? ~(uint64_t)0
                       : (((uint64_t)1 << (byte_len * 8)) - 1);
}

/* sign-extend a stored uint to int64 per load_data_to_datum
 * (ob_encoding_util.h:498-530): mask = ~INTEGER_MASK_TABLE[type_store_size] */
static inline uint64_t obx_sign_extend(uint64_t value, int64_t type_store_size,
                                       int is_signed_class) {
  if (!is_signed_class || type_store_size >= 8) return value;
  uint64_t mask = ~obx_integer_mask(type_store_size);
  if (value & (mask >> 1)) value |= mask;
  return value;
}

/* get_packing_size (ob_encoding_util.cpp:30-66), enable_bit_packing = true.
 * Returns bit width if *bit_packing, else byte width. */
static inline int64_t obx_packing_size(int *bit_packing, uint64_t v) {
  int64_t bit_size = (v == 0) ? 1 : (64 - __builtin_clzll(v));
  int64_t size = bit_size / 8;
  int64_t ext = bit_size % 8;
  if (ext == 0) {
    *bit_packing = 0;
  } else if (8 - ext < size / 2 + 1) {
    size++;
    *bit_packing = 0;
  } else {
    *bit_packing = 1;
  }
  return *bit_packing ? bit_size : size;
}

/* get_int_size (ob_encoding_util.cpp:68-75): minimal byte count */
static inline int64_t obx_int_size(uint64_t v) {
  int64_t bits = (v == 0) ? 1 : (64 - __builtin_clzll(v));
  return (bits + 7) / 8;
}

/* get_byte_packed_int_size (ob_encoding_util.cpp:77-89): 1/2/4/8 */
static inline int64_t obx_byte_packed_int_size(uint64_t v) {
  if (v <= 0xffu) return 1;
  if (v <= 0xffffu) return 2;
  if (v <= 0xffffffffu) return 4;
  return 8;
}

/* type store sizes (get_type_size_map, ob_encoding_util.h:128-180);
 * -1 = var length */
static inline int64_t obx_type_store_size(uint8_t obj_type) {
  switch (obj_type) {
    case 5: return 8;   /* ObIntType */
    case 4: return 4;   /* ObInt32Type */
    case 19: return 4;  /* ObDateType */
    case 23: return -1; /* ObCharType */
    case 50: return -1; /* ObDecimalIntType (ObDecimalIntSC, fixed datum len) */
    default: return -1;
  }
}

/* store class (get_store_class_map, ob_encoding_util.h:88-126) */
enum { OBX_SC_INT = 1, OBX_SC_UINT = 2, OBX_SC_DECIMAL = 4, OBX_SC_STRING = 5 };
static inline int obx_store_class(uint8_t obj_type) {
  switch (obj_type) {
    case 4: case 5: case 19: return OBX_SC_INT;
    case 23: return OBX_SC_STRING;
    case 50: return OBX_SC_DECIMAL;
    default: return 0;
  }
}

#ifdef __cplusplus
}
#endif
#endif /* OBX_FORMAT_H_ */
