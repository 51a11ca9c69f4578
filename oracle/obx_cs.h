/*
 * obx_cs.h — CS (column-store, OceanBase 4.3 "cs_encoding") format layer,
 * round-2 row of SURVEY §8(f): the INTEGER STREAM restatement.
 *
 * TEST INFRASTRUCTURE (oracle): restates, file-by-file,
 *   ob_stream_encoding_struct.{h,cpp} — ObIntegerStreamMeta field layout
 *     and its serialize format (i8 version/attr/type/width + conditional
 *     vi64 base / vi64 null_replace / i8 precision_width + i8
 *     pfor_packing_type for V2 metas, :20-95)
 *   deps/oblib/src/lib/utility/serialization.h:297 — encode_vi64: 7-bit
 *     little-endian groups with a 0x80 continuation bit (a negative value
 *     always serializes as 10 groups)
 *   ob_integer_stream_encoder.cpp:93-155 — datum→uint conversion (null →
 *     replace value, base subtraction FIRST) and RAW width-packed output
 *   ob_stream_encoding_struct.cpp:118-166 — build_signed_stream_meta's
 *     base/width selection: base = min ONLY when min < 0 (width covers
 *     range = max-min); otherwise no base, width covers max itself
 *
 * Scope of this slice: stream meta + RAW encoding type. The delta/zigzag/
 * PFoR codec types (ObIntegerStream::EncodingType 2-8, backed by the
 * deps/oblib ObCodec family) and the CS micro-block transformer/column
 * layer are the continuation rows.
 */
#ifndef OBX_CS_H_
#define OBX_CS_H_

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ObIntegerStream::Attribute */
enum {
  OBX_CS_USE_BASE = 0x1,
  OBX_CS_REPLACE_NULL = 0x2,
  OBX_CS_DECIMAL_INT = 0x4,
};

/* ObIntegerStream::EncodingType (subset implemented) */
enum {
  OBX_CS_ENC_RAW = 1,
  OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_RLE = 2,
  OBX_CS_ENC_DELTA_ZIGZAG_RLE = 4,
  OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_PFOR = 3,
  OBX_CS_ENC_DELTA_ZIGZAG_PFOR = 5,
  OBX_CS_ENC_SIMD_FIXEDPFOR = 6,
  OBX_CS_ENC_XOR_FIXED_PFOR = 8,
};

/* in-memory mirror of ObIntegerStreamMeta (the serialized form is
 * variable-length; see obx_cs_int_meta_enc) */
typedef struct obx_cs_int_meta {
  uint8_t version;   /* 1 = V2 (writes pfor_packing_type) */
  uint8_t attr;
  uint8_t type;      /* OBX_CS_ENC_* */
  uint8_t width_tag; /* 0/1/2/3 -> 1/2/4/8 bytes (UintWidth) */
  uint64_t base;
  uint64_t null_replaced;
  uint8_t precision_width_tag;
  uint8_t pfor_packing_type; /* 0 = CPU_ARCH_INDEPENDANT_SCALAR */
} obx_cs_int_meta;

/* OceanBase vi64 varint. Returns bytes written / read, or -1. */
int obx_cs_vi64_enc(uint8_t *buf, size_t cap, int64_t v);
int obx_cs_vi64_dec(const uint8_t *buf, size_t len, int64_t *out);

/* serialize / deserialize the stream meta; returns byte count or -1 */
int obx_cs_int_meta_enc(const obx_cs_int_meta *m, uint8_t *buf, size_t cap);
int obx_cs_int_meta_dec(const uint8_t *buf, size_t len, obx_cs_int_meta *m);

/* Encode a signed-int64 column slice as [meta][RAW width-packed stream].
 * nulls: optional bitmap (bit r set = NULL; replaced in-stream with the
 * base value, REPLACE_NULL semantics left to the column layer). Returns
 * total bytes or -1. */
int64_t obx_cs_int_stream_enc(const int64_t *vals, const uint8_t *nulls,
                              uint32_t rows, uint8_t *buf, size_t cap);

/* like obx_cs_int_stream_enc with an explicit stream encoding type
 * (OBX_CS_ENC_RAW or OBX_CS_ENC_DELTA_ZIGZAG_RLE) */
int64_t obx_cs_int_stream_enc2(const int64_t *vals, const uint8_t *nulls,
                               uint32_t rows, uint8_t enc_type,
                               uint8_t *buf, size_t cap);

/* full column-layer form: when use_null_replace is set, null rows are
 * stored as null_replaced (chosen by the caller adjacent to the data
 * range per ob_integer_column_encoder.cpp:190-220, so decoders recover
 * nulls by equality) and REPLACE_NULL + the vi64 value go in the meta */
int64_t obx_cs_int_stream_enc3(const int64_t *vals, const uint8_t *nulls,
                               uint32_t rows, uint8_t enc_type,
                               int use_null_replace, int64_t null_replaced,
                               uint8_t *buf, size_t cap);

/* enc3 plus the DECIMAL_INT attribute: precision_width_size in bytes
 * (4 or 8, 0 = not decimal) serialized as the precision width tag
 * (ObIntegerStreamMeta::set_precision_width_size,
 * ob_stream_encoding_struct.h) */
int64_t obx_cs_int_stream_enc4(const int64_t *vals, const uint8_t *nulls,
                               uint32_t rows, uint8_t enc_type,
                               int use_null_replace, int64_t null_replaced,
                               uint32_t precision_width_size,
                               uint8_t *buf, size_t cap);

/* Decode [meta][stream] back to int64 values (base re-applied). null
 * positions decode to the stored replace value. Returns consumed bytes
 * or -1. */
int64_t obx_cs_int_stream_dec(const uint8_t *buf, size_t len, uint32_t rows,
                              int64_t *out, obx_cs_int_meta *meta_out);

/* ---- string stream (ObStringStreamMeta, ob_stream_encoding_struct.h:341
 * and its DEFINE_SERIALIZE at .cpp:248: i8 version, i8 attr,
 * vi32 uncompressed_len, vi32 fixed_str_len when IS_FIXED_LEN_STRING).
 * The byte stream itself is the concatenated string bytes; var-length
 * columns pair it with a companion INTEGER stream of end offsets
 * (ObStringStreamEncoder), which callers build with
 * obx_cs_int_stream_enc. ---- */

enum {
  OBX_CS_STR_ZERO_LEN_NULL = 0x1,
  OBX_CS_STR_FIXED_LEN = 0x2,
};

typedef struct obx_cs_str_meta {
  uint8_t version;
  uint8_t attr;
  uint32_t uncompressed_len;
  uint32_t fixed_str_len;
} obx_cs_str_meta;

int obx_cs_vi32_enc(uint8_t *buf, size_t cap, int32_t v);
int obx_cs_vi32_dec(const uint8_t *buf, size_t len, int32_t *out);
int obx_cs_str_meta_enc(const obx_cs_str_meta *m, uint8_t *buf, size_t cap);
int obx_cs_str_meta_dec(const uint8_t *buf, size_t len, obx_cs_str_meta *m);

/* Fixed-length string stream: [meta][rows*fixed_len bytes]. Returns total
 * bytes or -1. Decode verifies the meta and returns a pointer offset. */
int64_t obx_cs_str_stream_enc_fixed(const uint8_t *bytes, uint32_t rows,
                                    uint32_t fixed_len, uint8_t *buf,
                                    size_t cap);
int64_t obx_cs_str_stream_dec_fixed(const uint8_t *buf, size_t len,
                                    uint32_t rows, uint32_t *fixed_len_out,
                                    const uint8_t **bytes_out);

/* DELTA_ZIGZAG_RLE codec (ObDeltaZigzagRleInner; see obx_cs.c header
 * comment for the cited bit protocol). in/out are width-packed arrays
 * (wb in {1,2,4,8}); returns encoded / consumed bytes or -1. */
int64_t obx_cs_dzr_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                       uint8_t *out, size_t cap);
int64_t obx_cs_dzr_dec(const uint8_t *in, size_t in_len, uint32_t count,
                       uint32_t wb, uint8_t *out);
int64_t obx_cs_ddzr_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                        uint8_t *out, size_t cap);
int64_t obx_cs_ddzr_dec(const uint8_t *in, size_t in_len, uint32_t count,
                        uint32_t wb, uint8_t *out);
/* DELTA_ZIGZAG_PFOR (128-value PFoR frames + SimpleBitPacking tail) */
int64_t obx_cs_dzp_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                       uint8_t *out, size_t cap);
int64_t obx_cs_dzp_dec(const uint8_t *in, size_t in_len, uint32_t count,
                       uint32_t wb, uint8_t *out);
int64_t obx_cs_ddzp_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                        uint8_t *out, size_t cap);
int64_t obx_cs_ddzp_dec(const uint8_t *in, size_t in_len, uint32_t count,
                        uint32_t wb, uint8_t *out);
int64_t obx_cs_fpfor_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                         uint8_t *out, size_t cap);
int64_t obx_cs_fpfor_dec(const uint8_t *in, size_t in_len, uint32_t count,
                         uint32_t wb, uint8_t *out);
int64_t obx_cs_xpfor_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                         uint8_t *out, size_t cap);
int64_t obx_cs_xpfor_dec(const uint8_t *in, size_t in_len, uint32_t count,
                         uint32_t wb, uint8_t *out);

#ifdef __cplusplus
}
#endif
#endif /* OBX_CS_H_ */
