/* CS integer-stream restatement — see obx_cs.h for the cited reference
 * functions. Oracle/test infrastructure only. */
#include "obx_cs.h"

#include <string.h>
#include <stdlib.h>

int obx_cs_vi64_enc(uint8_t *buf, size_t cap, int64_t v) {
  /* serialization::encode_vi64: 7-bit LE groups, 0x80 continuation */
  uint64_t u = (uint64_t)v;
  int n = 0;
  while (u > 0x7f) {
    if ((size_t)n >= cap) return -1;
    buf[n++] = (uint8_t)(u | 0x80);
    u >>= 7;
  }
  if ((size_t)n >= cap) return -1;
  buf[n++] = (uint8_t)(u & 0x7f);
  return n;
}

int obx_cs_vi64_dec(const uint8_t *buf, size_t len, int64_t *out) {
  uint64_t u = 0;
  int shift = 0, n = 0;
  for (;;) {
    if ((size_t)n >= len || shift > 63) return -1;
    uint8_t b = buf[n++];
    u |= (uint64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  *out = (int64_t)u;
  return n;
}

static const uint32_t WIDTH_BYTES[4] = {1, 2, 4, 8};

int obx_cs_int_meta_enc(const obx_cs_int_meta *m, uint8_t *buf, size_t cap) {
  size_t pos = 0;
  if (cap < 4) return -1;
  buf[pos++] = m->version;
  buf[pos++] = m->attr;
  buf[pos++] = m->type;
  buf[pos++] = m->width_tag;
  if (m->attr & OBX_CS_USE_BASE) {
    int n = obx_cs_vi64_enc(buf + pos, cap - pos, (int64_t)m->base);
    if (n < 0) return -1;
    pos += n;
  }
  if (m->attr & OBX_CS_REPLACE_NULL) {
    int n = obx_cs_vi64_enc(buf + pos, cap - pos, (int64_t)m->null_replaced);
    if (n < 0) return -1;
    pos += n;
  }
  if (m->attr & OBX_CS_DECIMAL_INT) {
    if (pos >= cap) return -1;
    buf[pos++] = m->precision_width_tag;
  }
  if (m->version > 0) { /* V2: pfor packing type */
    if (pos >= cap) return -1;
    buf[pos++] = m->pfor_packing_type;
  }
  return (int)pos;
}

int obx_cs_int_meta_dec(const uint8_t *buf, size_t len, obx_cs_int_meta *m) {
  size_t pos = 0;
  memset(m, 0, sizeof(*m));
  if (len < 4) return -1;
  m->version = buf[pos++];
  m->attr = buf[pos++];
  m->type = buf[pos++];
  m->width_tag = buf[pos++];
  if (m->width_tag > 3) return -1; /* is_valid: width <= UW_8_BYTE */
  if (m->attr & OBX_CS_USE_BASE) {
    int64_t v;
    int n = obx_cs_vi64_dec(buf + pos, len - pos, &v);
    if (n < 0) return -1;
    m->base = (uint64_t)v;
    pos += n;
  }
  if (m->attr & OBX_CS_REPLACE_NULL) {
    int64_t v;
    int n = obx_cs_vi64_dec(buf + pos, len - pos, &v);
    if (n < 0) return -1;
    m->null_replaced = (uint64_t)v;
    pos += n;
  }
  if (m->attr & OBX_CS_DECIMAL_INT) {
    if (pos >= len) return -1;
    m->precision_width_tag = buf[pos++];
  }
  if (m->version > 0) {
    if (pos >= len) return -1;
    m->pfor_packing_type = buf[pos++];
  } /* V1 implies CPU_ARCH_DEPENDANT; not produced by this writer */
  return (int)pos;
}

static inline int null_at(const uint8_t *nulls, uint32_t r) {
  return nulls && ((nulls[r >> 3] >> (r & 7)) & 1);
}

int64_t obx_cs_int_stream_enc(const int64_t *vals, const uint8_t *nulls,
                              uint32_t rows, uint8_t *buf, size_t cap) {
  return obx_cs_int_stream_enc2(vals, nulls, rows, OBX_CS_ENC_RAW, buf,
                                cap);
}

int64_t obx_cs_dzr_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                       uint8_t *out, size_t cap);

int64_t obx_cs_int_stream_enc2(const int64_t *vals, const uint8_t *nulls,
                               uint32_t rows, uint8_t enc_type,
                               uint8_t *buf, size_t cap) {
  return obx_cs_int_stream_enc3(vals, nulls, rows, enc_type, 0, 0, buf,
                                cap);
}

int64_t obx_cs_int_stream_enc3(const int64_t *vals, const uint8_t *nulls,
                               uint32_t rows, uint8_t enc_type,
                               int use_null_replace, int64_t null_replaced,
                               uint8_t *buf, size_t cap) {
  return obx_cs_int_stream_enc4(vals, nulls, rows, enc_type,
                                use_null_replace, null_replaced, 0, buf,
                                cap);
}

int64_t obx_cs_int_stream_enc4(const int64_t *vals, const uint8_t *nulls,
                               uint32_t rows, uint8_t enc_type,
                               int use_null_replace, int64_t null_replaced,
                               uint32_t precision_width_size,
                               uint8_t *buf, size_t cap) {
  if (!vals || !rows || !buf) return -1;
  /* build_signed_stream_meta (ob_stream_encoding_struct.cpp:118-166):
     base ONLY when min < 0 (range = max - min); for min >= 0 the width
     covers max itself and no base is stored. When the column layer
     chose a null-replace value it already sits adjacent to the range
     (min-1 or max+1, ob_integer_column_encoder.cpp:190-220), so folding
     it into the min/max scan reproduces new_int_min/new_int_max. */
  int64_t mn = 0, mx = 0;
  int any = 0;
  for (uint32_t r = 0; r < rows; r++) {
    if (null_at(nulls, r)) continue;
    if (!any || vals[r] < mn) mn = vals[r];
    if (!any || vals[r] > mx) mx = vals[r];
    any = 1;
  }
  if (use_null_replace) {
    if (!any || null_replaced < mn) mn = null_replaced;
    if (!any || null_replaced > mx) mx = null_replaced;
    any = 1;
  }
  if (!any) mn = mx = 0;
  obx_cs_int_meta m;
  memset(&m, 0, sizeof(m));
  m.version = 1; /* V2 */
  m.type = enc_type;
  uint64_t range;
  if (mn < 0) {
    m.attr = OBX_CS_USE_BASE;
    m.base = (uint64_t)mn;
    range = (uint64_t)mx - (uint64_t)mn;
  } else {
    m.attr = 0;
    m.base = 0;
    range = (uint64_t)mx;
  }
  if (use_null_replace) {
    m.attr |= OBX_CS_REPLACE_NULL;
    m.null_replaced = (uint64_t)null_replaced;
  }
  if (precision_width_size) {
    /* set_precision_width_size: tag of the decimal datum width */
    m.attr |= OBX_CS_DECIMAL_INT;
    m.precision_width_tag = precision_width_size <= 1 ? 0
                            : precision_width_size <= 2 ? 1
                            : precision_width_size <= 4 ? 2 : 3;
  }
  m.width_tag = range <= 0xFF ? 0 : range <= 0xFFFF ? 1
                : range <= 0xFFFFFFFFull ? 2 : 3;
  m.pfor_packing_type = 0; /* CPU_ARCH_INDEPENDANT_SCALAR */
  int hn = obx_cs_int_meta_enc(&m, buf, cap);
  if (hn < 0) return -1;
  size_t pos = (size_t)hn;
  uint32_t wb = WIDTH_BYTES[m.width_tag];
  if (pos + (size_t)rows * wb + 32 > cap) return -1;
  /* datum->uint conversion (ob_integer_stream_encoder.cpp:105-113):
     null -> replace value if set, else base (else 0); base subtracted
     after */
  uint64_t null_fill = use_null_replace ? (uint64_t)null_replaced : m.base;
  uint8_t *packed = buf + pos; /* RAW writes in place */
  uint8_t *tmp = NULL;
  if (enc_type != OBX_CS_ENC_RAW) {
    tmp = (uint8_t *)malloc((size_t)rows * wb);
    if (!tmp) return -1;
    packed = tmp;
  }
  for (uint32_t r = 0; r < rows; r++) {
    uint64_t ele = null_at(nulls, r) ? null_fill : (uint64_t)vals[r];
    ele -= m.base;
    memcpy(packed + (size_t)r * wb, &ele, wb);
  }
  if (enc_type == OBX_CS_ENC_RAW) {
    pos += (size_t)rows * wb;
  } else if (enc_type == OBX_CS_ENC_DELTA_ZIGZAG_RLE ||
             enc_type == OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_RLE ||
             enc_type == OBX_CS_ENC_DELTA_ZIGZAG_PFOR ||
             enc_type == OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_PFOR ||
             enc_type == OBX_CS_ENC_SIMD_FIXEDPFOR ||
             enc_type == OBX_CS_ENC_XOR_FIXED_PFOR) {
    int64_t n = enc_type == OBX_CS_ENC_DELTA_ZIGZAG_RLE
                    ? obx_cs_dzr_enc(packed, rows, wb, buf + pos, cap - pos)
                : enc_type == OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_RLE
                    ? obx_cs_ddzr_enc(packed, rows, wb, buf + pos,
                                      cap - pos)
                : enc_type == OBX_CS_ENC_DELTA_ZIGZAG_PFOR
                    ? obx_cs_dzp_enc(packed, rows, wb, buf + pos,
                                     cap - pos)
                : enc_type == OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_PFOR
                    ? obx_cs_ddzp_enc(packed, rows, wb, buf + pos,
                                      cap - pos)
                : enc_type == OBX_CS_ENC_SIMD_FIXEDPFOR
                    ? obx_cs_fpfor_enc(packed, rows, wb, buf + pos,
                                       cap - pos)
                    : obx_cs_xpfor_enc(packed, rows, wb, buf + pos,
                                       cap - pos);
    free(tmp);
    if (n < 0) return -1;
    pos += (size_t)n;
  } else {
    free(tmp);
    return -1;
  }
  return (int64_t)pos;
}

int64_t obx_cs_int_stream_dec(const uint8_t *buf, size_t len, uint32_t rows,
                              int64_t *out, obx_cs_int_meta *meta_out) {
  obx_cs_int_meta m;
  int hn = obx_cs_int_meta_dec(buf, len, &m);
  if (hn < 0) return -1;
  size_t pos = (size_t)hn;
  uint32_t wb = WIDTH_BYTES[m.width_tag];
  uint64_t base = (m.attr & OBX_CS_USE_BASE) ? m.base : 0;
  if (m.type == OBX_CS_ENC_RAW) {
    if (pos + (size_t)rows * wb > len) return -1;
    for (uint32_t r = 0; r < rows; r++) {
      uint64_t ele = 0;
      memcpy(&ele, buf + pos, wb);
      pos += wb;
      out[r] = (int64_t)(ele + base);
    }
  } else if (m.type == OBX_CS_ENC_DELTA_ZIGZAG_RLE ||
             m.type == OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_RLE ||
             m.type == OBX_CS_ENC_DELTA_ZIGZAG_PFOR ||
             m.type == OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_PFOR ||
             m.type == OBX_CS_ENC_SIMD_FIXEDPFOR ||
             m.type == OBX_CS_ENC_XOR_FIXED_PFOR) {
    uint8_t *packed = (uint8_t *)malloc((size_t)rows * wb);
    if (!packed) return -1;
    int64_t n = m.type == OBX_CS_ENC_DELTA_ZIGZAG_RLE
                    ? obx_cs_dzr_dec(buf + pos, len - pos, rows, wb, packed)
                : m.type == OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_RLE
                    ? obx_cs_ddzr_dec(buf + pos, len - pos, rows, wb,
                                      packed)
                : m.type == OBX_CS_ENC_DELTA_ZIGZAG_PFOR
                    ? obx_cs_dzp_dec(buf + pos, len - pos, rows, wb, packed)
                : m.type == OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_PFOR
                    ? obx_cs_ddzp_dec(buf + pos, len - pos, rows, wb,
                                      packed)
                : m.type == OBX_CS_ENC_SIMD_FIXEDPFOR
                    ? obx_cs_fpfor_dec(buf + pos, len - pos, rows, wb,
                                       packed)
                    : obx_cs_xpfor_dec(buf + pos, len - pos, rows, wb,
                                       packed);
    if (n < 0) { free(packed); return -1; }
    pos += (size_t)n;
    for (uint32_t r = 0; r < rows; r++) {
      uint64_t ele = 0;
      memcpy(&ele, packed + (size_t)r * wb, wb);
      out[r] = (int64_t)(ele + base);
    }
    free(packed);
  } else {
    return -1;
  }
  if (meta_out) *meta_out = m;
  return (int64_t)pos;
}

int obx_cs_vi32_enc(uint8_t *buf, size_t cap, int32_t v) {
  uint32_t u = (uint32_t)v;
  int n = 0;
  while (u > 0x7f) {
    if ((size_t)n >= cap) return -1;
    buf[n++] = (uint8_t)(u | 0x80);
    u >>= 7;
  }
  if ((size_t)n >= cap) return -1;
  buf[n++] = (uint8_t)(u & 0x7f);
  return n;
}

int obx_cs_vi32_dec(const uint8_t *buf, size_t len, int32_t *out) {
  uint32_t u = 0;
  int shift = 0, n = 0;
  for (;;) {
    if ((size_t)n >= len || shift > 31) return -1;
    uint8_t b = buf[n++];
    u |= (uint32_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  *out = (int32_t)u;
  return n;
}

int obx_cs_str_meta_enc(const obx_cs_str_meta *m, uint8_t *buf, size_t cap) {
  size_t pos = 0;
  if (cap < 2) return -1;
  buf[pos++] = m->version;
  buf[pos++] = m->attr;
  int n = obx_cs_vi32_enc(buf + pos, cap - pos,
                          (int32_t)m->uncompressed_len);
  if (n < 0) return -1;
  pos += n;
  if (m->attr & OBX_CS_STR_FIXED_LEN) {
    n = obx_cs_vi32_enc(buf + pos, cap - pos, (int32_t)m->fixed_str_len);
    if (n < 0) return -1;
    pos += n;
  }
  return (int)pos;
}

int obx_cs_str_meta_dec(const uint8_t *buf, size_t len, obx_cs_str_meta *m) {
  size_t pos = 0;
  memset(m, 0, sizeof(*m));
  if (len < 2) return -1;
  m->version = buf[pos++];
  m->attr = buf[pos++];
  int32_t v;
  int n = obx_cs_vi32_dec(buf + pos, len - pos, &v);
  if (n < 0) return -1;
  m->uncompressed_len = (uint32_t)v;
  pos += n;
  if (m->attr & OBX_CS_STR_FIXED_LEN) {
    n = obx_cs_vi32_dec(buf + pos, len - pos, &v);
    if (n < 0) return -1;
    m->fixed_str_len = (uint32_t)v;
    pos += n;
  }
  return (int)pos;
}

int64_t obx_cs_str_stream_enc_fixed(const uint8_t *bytes, uint32_t rows,
                                    uint32_t fixed_len, uint8_t *buf,
                                    size_t cap) {
  obx_cs_str_meta m;
  memset(&m, 0, sizeof(m));
  m.version = 0;
  m.attr = OBX_CS_STR_FIXED_LEN;
  m.uncompressed_len = rows * fixed_len;
  m.fixed_str_len = fixed_len;
  int hn = obx_cs_str_meta_enc(&m, buf, cap);
  if (hn < 0) return -1;
  size_t pos = (size_t)hn;
  if (pos + (size_t)rows * fixed_len > cap) return -1;
  memcpy(buf + pos, bytes, (size_t)rows * fixed_len);
  return (int64_t)(pos + (size_t)rows * fixed_len);
}

int64_t obx_cs_str_stream_dec_fixed(const uint8_t *buf, size_t len,
                                    uint32_t rows, uint32_t *fixed_len_out,
                                    const uint8_t **bytes_out) {
  obx_cs_str_meta m;
  int hn = obx_cs_str_meta_dec(buf, len, &m);
  if (hn < 0 || !(m.attr & OBX_CS_STR_FIXED_LEN)) return -1;
  if (m.uncompressed_len != rows * m.fixed_str_len) return -1;
  if ((size_t)hn + m.uncompressed_len > len) return -1;
  *fixed_len_out = m.fixed_str_len;
  *bytes_out = buf + hn;
  return (int64_t)hn + m.uncompressed_len;
}

/* ====================================================================== */
/* DELTA_ZIGZAG_RLE codec (ObDeltaZigzagRleInner, deps/oblib/src/lib/
 * codec/ob_delta_zigzag_rle.h:22-318, over the ObBitUtils protocol of
 * ob_bp_util.h:128-340 and ObZigZag:74). Bit stream semantics mirrored
 * exactly: little-endian bit accumulation (bw += v << br), e_slide
 * writes 8 raw bytes and advances br/8, put<u64> splits >45-bit values
 * as (high width-32, slide, low 32), flush advances ceil(br/8).
 * Repeat runs: 1..18 as single '1' bits; longer as
 * [4b zero flag][3b zero][3b byte-cnt-1][byte-cnt*8b of cnt-18].
 * Deltas (zigzag at element width): 0 -> nothing; < 2^(N2-1) ->
 * [delta<<2|0b10] in N2+2 bits; N3/N4 likewise with 0b100/0b1000; else
 * [4b zero][3b byte-cnt-1][byte-cnt*8b delta]. */

static const uint32_t DZR_N2[4] = {3, 6, 6, 6};
static const uint32_t DZR_N3[4] = {5, 12, 10, 12};
static const uint32_t DZR_N4[4] = {9, 17, 17, 20};
#define DZR_BASE_REPEAT 18

typedef struct {
  uint64_t bw;
  uint32_t br;
  uint8_t *op;
  uint8_t *end;
} dzr_e;

static inline int dzr_e_room(dzr_e *e, size_t n) {
  return (size_t)(e->end - e->op) >= n + 16;
}
static inline void dzr_put(dzr_e *e, uint32_t width, uint64_t v) {
  e->bw += v << e->br;
  e->br += width;
}
static inline void dzr_slide(dzr_e *e) {
  memcpy(e->op, &e->bw, 8);
  e->op += e->br >> 3;
  e->bw >>= (e->br & ~7u);
  e->br &= 7;
}
static inline void dzr_put64(dzr_e *e, uint32_t width, uint64_t v) {
  if (width > 45) { /* put<uint64_t> split, ob_bp_util.h:300-310 */
    dzr_put(e, width - 32, v >> 32);
    dzr_slide(e);
    dzr_put(e, 32, (uint32_t)v);
  } else {
    dzr_put(e, width, v);
  }
}

static inline uint32_t dzr_bits_u64(uint64_t v) { /* gccbits */
  uint32_t n = 0;
  while (v) { n++; v >>= 1; }
  return n ? n : 1;
}

static inline uint64_t dzr_zz_enc(uint64_t v, uint32_t wbits) {
  uint64_t m = wbits >= 64 ? ~0ull : ((1ull << wbits) - 1);
  v &= m;
  uint64_t sign = (v >> (wbits - 1)) & 1;
  return (((v << 1) ^ (sign ? m : 0)) & m);
}
static inline uint64_t dzr_zz_dec(uint64_t v, uint32_t wbits) {
  uint64_t m = wbits >= 64 ? ~0ull : ((1ull << wbits) - 1);
  v &= m;
  return (((v >> 1) ^ (0ull - (v & 1))) & m);
}

static inline uint64_t dzr_load(const uint8_t *p, uint32_t wb) {
  uint64_t v = 0;
  memcpy(&v, p, wb);
  return v;
}

/* emit pending repeat count + one (nonzero or final-zero) delta */
static int dzr_emit(dzr_e *e, uint64_t r, uint64_t delta_zz, int wi) {
  if (!dzr_e_room(e, 32)) return -1;
  if (r > DZR_BASE_REPEAT) {
    r -= DZR_BASE_REPEAT;
    uint32_t b = (dzr_bits_u64(r) + 7) >> 3;
    dzr_put(e, 4 + 3 + 3, (uint64_t)(b - 1) << (4 + 3));
    dzr_put64(e, b << 3, r);
    dzr_slide(e);
  } else {
    while (r--) {
      dzr_put(e, 1, 1);
      dzr_slide(e);
    }
  }
  const uint32_t N2 = DZR_N2[wi], N3 = DZR_N3[wi], N4 = DZR_N4[wi];
  if (delta_zz == 0) {
    /* nothing: only legal as the final element (decode stops by count) */
  } else if (delta_zz < (1ull << (N2 - 1))) {
    dzr_put(e, N2 + 2, (delta_zz << 2) | 2);
  } else if (delta_zz < (1ull << (N3 - 1))) {
    dzr_put(e, N3 + 3, (delta_zz << 3) | 4);
  } else if (delta_zz < (1ull << (N4 - 1))) {
    dzr_put(e, N4 + 4, (delta_zz << 4) | 8);
  } else {
    uint32_t b = (dzr_bits_u64(delta_zz) + 7) >> 3;
    dzr_put(e, 4 + 3, (uint64_t)(b - 1) << 4);
    dzr_put64(e, b << 3, delta_zz);
  }
  dzr_slide(e);
  return 0;
}

/* order 1 = DELTA_ZIGZAG_RLE; order 2 = DOUBLE_DELTA_ZIGZAG_RLE
   (ob_double_delta_zigzag_rle.h: dd = delta - prev_delta; repeat runs
   are constant-SLOPE spans, decoded as start += pd each) */
static int64_t dzr_enc_core(const uint8_t *in, uint32_t count, uint32_t wb,
                            uint8_t *out, size_t cap, int order) {
  int wi = wb == 1 ? 0 : wb == 2 ? 1 : wb == 4 ? 2 : 3;
  uint32_t wbits = wb * 8;
  uint64_t wmask = wbits >= 64 ? ~0ull : ((1ull << wbits) - 1);
  dzr_e e = {0, 0, out, out + cap};
  if (cap < 24) return -1;
  uint64_t start = 0, pd = 0, pending = 0;
  for (uint32_t i = 0; i < count; i++) {
    uint64_t v = dzr_load(in + (size_t)i * wb, wb);
    uint64_t d1 = (v - start) & wmask;
    start = v;
    uint64_t dd = d1;
    if (order == 2) {
      dd = (d1 - pd) & wmask;
      pd = d1;
    }
    if (dd == 0) {
      pending++;
      continue;
    }
    if (dzr_emit(&e, pending, dzr_zz_enc(dd, wbits), wi)) return -1;
    pending = 0;
  }
  if (pending) {
    if (dzr_emit(&e, pending, 0, wi)) return -1;
  }
  /* flush (ob_bp_util.h:236): write bw, advance ceil(br/8) */
  if (!dzr_e_room(&e, 8)) return -1;
  memcpy(e.op, &e.bw, 8);
  e.op += (e.br + 7) >> 3;
  return (int64_t)(e.op - out);
}

int64_t obx_cs_dzr_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                       uint8_t *out, size_t cap) {
  return dzr_enc_core(in, count, wb, out, cap, 1);
}

int64_t obx_cs_ddzr_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                        uint8_t *out, size_t cap) {
  return dzr_enc_core(in, count, wb, out, cap, 2);
}

typedef struct {
  uint64_t bw;
  uint32_t br;
  const uint8_t *ip;
  const uint8_t *end;
} dzr_d;

static inline void dzr_d_slide(dzr_d *d) {
  d->ip += d->br >> 3;
  if (d->ip + 8 <= d->end) {
    memcpy(&d->bw, d->ip, 8);
  } else {
    uint8_t tmp[8] = {0};
    if (d->ip < d->end) memcpy(tmp, d->ip, (size_t)(d->end - d->ip));
    memcpy(&d->bw, tmp, 8);
  }
  d->br &= 7;
}
static inline uint64_t dzr_get(dzr_d *d, uint32_t b) { /* get57 semantics */
  uint64_t v = (d->bw >> d->br);
  v &= (b >= 64) ? ~0ull : ((1ull << b) - 1);
  d->br += b;
  return v;
}

static int64_t dzr_dec_core(const uint8_t *in, size_t in_len,
                            uint32_t count, uint32_t wb, uint8_t *out,
                            int order) {
  int wi = wb == 1 ? 0 : wb == 2 ? 1 : wb == 4 ? 2 : 3;
  uint32_t wbits = wb * 8;
  uint64_t wmask = wbits >= 64 ? ~0ull : ((1ull << wbits) - 1);
  const uint32_t N2 = DZR_N2[wi], N3 = DZR_N3[wi], N4 = DZR_N4[wi];
  dzr_d d = {0, 0, in, in + in_len};
  uint64_t start = 0, pd = 0;
  uint32_t done = 0;
  dzr_d_slide(&d);
  while (done < count) {
    uint64_t peek = d.bw >> d.br;
    uint64_t delta;
    if (peek & 1) {
      d.br += 1;
      delta = 0;
    } else if (peek & 2) {
      d.br += N2 + 2;
      delta = (peek >> 2) & ((1ull << N2) - 1);
    } else if (peek & 4) {
      d.br += N3 + 3;
      delta = (peek >> 3) & ((1ull << N3) - 1);
    } else if (peek & 8) {
      d.br += N4 + 4;
      delta = (peek >> 4) & ((1ull << N4) - 1);
    } else {
      uint32_t f = (uint32_t)dzr_get(&d, 4 + 3);
      uint32_t b = f >> 4;
      if (b == 0) { /* long repeat count */
        b = (uint32_t)dzr_get(&d, 3);
        uint64_t r = dzr_get(&d, (b + 1) << 3); /* get<uint32_t>=get57 */
        dzr_d_slide(&d);
        r += DZR_BASE_REPEAT;
        while (r-- && done < count) {
          if (order == 2) start = (start + pd) & wmask;
          memcpy(out + (size_t)done * wb, &start, wb);
          done++;
        }
        continue;
      }
      if (b == 1) return -1; /* overflow: "can not be" */
      uint32_t bits = (b + 1) << 3;
      if (wb == 8 && bits > 45) { /* get<uint64_t> split */
        uint64_t hi = dzr_get(&d, bits - 32);
        dzr_d_slide(&d);
        uint64_t lo = dzr_get(&d, 32);
        delta = (hi << 32) | lo;
      } else {
        delta = dzr_get(&d, bits);
      }
    }
    if (order == 2) {
      pd = (pd + dzr_zz_dec(delta, wbits)) & wmask;
      start = (start + pd) & wmask;
    } else {
      start = (start + dzr_zz_dec(delta, wbits)) & wmask;
    }
    memcpy(out + (size_t)done * wb, &start, wb);
    done++;
    dzr_d_slide(&d);
  }
  /* align (ob_bp_util.h:152) */
  d.ip += (d.br + 7) >> 3;
  return (int64_t)(d.ip - in);
}

int64_t obx_cs_dzr_dec(const uint8_t *in, size_t in_len, uint32_t count,
                       uint32_t wb, uint8_t *out) {
  return dzr_dec_core(in, in_len, count, wb, out, 1);
}

int64_t obx_cs_ddzr_dec(const uint8_t *in, size_t in_len, uint32_t count,
                        uint32_t wb, uint8_t *out) {
  return dzr_dec_core(in, in_len, count, wb, out, 2);
}

/* ====================================================================== */
/* DELTA_ZIGZAG_PFOR codec (ObDeltaZigzagFixedPfor, ob_delta_zigzag_pfor.h
 * + ObSIMDFixedPFor::__encode_array, ob_simd_fixed_pfor.h:30-200, in the
 * CPU_ARCH_INDEPENDANT_SCALAR layout = flat LSB-first bit packing,
 * ob_bp_helpers.h:1378 scalar_bit_packing over the generated sequential
 * pack layout). Frame: 128-value blocks of zigzag deltas; per block the
 * cost model (find_most_fit_bx) picks a low width b with exceptions:
 * header [b] (no exceptions) or [0x80|b][bx]; with exceptions the block
 * is [16-byte exception bitmap][exceptions' high parts (v>>b) packed at
 * bx bits, byte-rounded][all values' low b bits packed]. The tail
 * (<128 values) is ObSimpleBitPacking: [max-bits byte][flat packed]. */

#define DZP_BLOCK 128

static inline uint32_t dzp_bits0(uint64_t v) { /* gccbits: 0 for 0 */
  uint32_t n = 0;
  while (v) { n++; v >>= 1; }
  return n;
}

/* flat LSB-first bit packing, byte-rounded at the end (up to 71 bits per
   element span handled through a 128-bit accumulator; oracle clarity
   over speed) */
static size_t dzp_pack(const uint64_t *v, uint32_t n, uint32_t b,
                       uint8_t *out) {
  if (b == 0) return 0;
  size_t total = ((size_t)n * b + 7) / 8;
  memset(out, 0, total + 8);
  uint64_t bitpos = 0;
  for (uint32_t i = 0; i < n; i++) {
    uint64_t val = b >= 64 ? v[i] : (v[i] & ((1ull << b) - 1));
    size_t byte = bitpos >> 3;
    uint32_t sh = (uint32_t)(bitpos & 7);
    unsigned __int128 acc = (unsigned __int128)val << sh;
    for (uint32_t k = 0; k * 8 < sh + b; k++)
      out[byte + k] |= (uint8_t)(acc >> (8 * k));
    bitpos += b;
  }
  return total;
}

static void dzp_unpack(const uint8_t *in, uint32_t n, uint32_t b,
                       uint64_t *out) {
  if (b == 0) {
    memset(out, 0, (size_t)n * 8);
    return;
  }
  uint64_t bitpos = 0;
  for (uint32_t i = 0; i < n; i++) {
    size_t byte = bitpos >> 3;
    uint32_t sh = (uint32_t)(bitpos & 7);
    unsigned __int128 acc = 0;
    for (uint32_t k = 0; k * 8 < sh + b; k++)
      acc |= (unsigned __int128)in[byte + k] << (8 * k);
    uint64_t v = (uint64_t)(acc >> sh);
    if (b < 64) v &= (1ull << b) - 1;
    out[i] = v;
    bitpos += b;
  }
}

/* bounds guard for truncated/hostile streams: room for n b-bit values */
static inline int dzp_room(const uint8_t *ip, const uint8_t *end,
                           uint32_t n, uint32_t b) {
  return b <= 64 && ip <= end &&
         (size_t)(end - ip) >= ((size_t)n * b + 7) / 8;
}

/* find_most_fit_bx (ob_simd_fixed_pfor.h:30-87) */
static void dzp_fit(const uint64_t *v, uint32_t n, uint32_t wbits,
                    uint32_t *b_out, uint32_t *bx_out) {
  uint32_t cnt[65] = {0};
  uint64_t u = 0;
  for (uint32_t i = 0; i < n; i++) {
    cnt[dzp_bits0(v[i])]++;
    u |= v[i];
  }
  int32_t b = (int32_t)dzp_bits0(u);
  uint32_t bx = (uint32_t)b;
  int32_t ml = (int32_t)(((uint64_t)n * b + 7) / 8) + 1;
  uint32_t x = cnt[b];
  uint32_t bmp8 = (n + 7) / 8;
  for (int32_t i = b - 1; i >= 0; --i) {
    int32_t l = (int32_t)(2 + bmp8 + (((uint64_t)x * (bx - i) + 7) / 8) +
                          (((uint64_t)n * i + 7) / 8));
    x += cnt[i];
    if (l < ml) {
      ml = l;
      b = i;
    }
  }
  *b_out = (uint32_t)b;
  *bx_out = bx - (uint32_t)b;
  (void)wbits;
}

/* transform: 1 = delta+zigzag (DELTA_ZIGZAG_PFOR), 2 = double-delta+
   zigzag (DOUBLE_DELTA_ZIGZAG_PFOR), 0 = none (SIMD_FIXEDPFOR; the
   reference frames EVERYTHING in 128s over padded buffers — our
   restatement zero-pads the final partial frame, documented divergence
   from the in-place over-read) */
static int64_t dzp_enc_core(const uint8_t *in, uint32_t count, uint32_t wb,
                            uint8_t *out, size_t cap, int transform) {
  uint32_t wbits = wb * 8;
  uint64_t wmask = wbits >= 64 ? ~0ull : ((1ull << wbits) - 1);
  uint8_t *op = out;
  uint8_t *end = out + cap;
  uint64_t start = 0, pd = 0;
  uint64_t zz[DZP_BLOCK];
  uint32_t done = 0;
  while (count - done >= DZP_BLOCK) {
    for (uint32_t i = 0; i < DZP_BLOCK; i++) {
      uint64_t v = dzr_load(in + (size_t)(done + i) * wb, wb);
      if (transform == 0) {
        zz[i] = v;
      } else if (transform == 1) {
        zz[i] = dzr_zz_enc((v - start) & wmask, wbits);
        start = v;
      } else {
        uint64_t d1 = (v - start) & wmask;
        start = v;
        zz[i] = dzr_zz_enc((d1 - pd) & wmask, wbits);
        pd = d1;
      }
    }
    uint32_t b, bx;
    dzp_fit(zz, DZP_BLOCK, wbits, &b, &bx);
    if ((size_t)(end - op) < 2 + 16 + (size_t)DZP_BLOCK * wb + 24)
      return -1;
    if (bx == 0) {
      *op++ = (uint8_t)b;
      op += dzp_pack(zz, DZP_BLOCK, b, op);
    } else {
      *op++ = (uint8_t)(0x80 | b);
      *op++ = (uint8_t)bx;
      uint64_t msk = b >= 64 ? ~0ull : ((1ull << b) - 1);
      uint64_t lowv[DZP_BLOCK], exc[DZP_BLOCK];
      uint64_t xmap[DZP_BLOCK / 64] = {0, 0};
      uint32_t xn = 0;
      for (uint32_t i = 0; i < DZP_BLOCK; i++) {
        lowv[i] = zz[i] & msk;
        if (zz[i] > msk) {
          xmap[i >> 6] |= 1ull << (i & 63);
          exc[xn++] = zz[i] >> b;
        }
      }
      memcpy(op, xmap, DZP_BLOCK / 8);
      op += DZP_BLOCK / 8;
      op += dzp_pack(exc, xn, bx, op);
      op += dzp_pack(lowv, DZP_BLOCK, b, op);
    }
    done += DZP_BLOCK;
  }
  if (done < count) { /* tail: ObSimpleBitPacking (1-byte maxbits) */
    uint32_t rem = count - done;
    uint64_t u = 0;
    for (uint32_t i = 0; i < rem; i++) {
      uint64_t v = dzr_load(in + (size_t)(done + i) * wb, wb);
      if (transform == 0) {
        zz[i] = v;
      } else if (transform == 1) {
        zz[i] = dzr_zz_enc((v - start) & wmask, wbits);
        start = v;
      } else {
        uint64_t d1 = (v - start) & wmask;
        start = v;
        zz[i] = dzr_zz_enc((d1 - pd) & wmask, wbits);
        pd = d1;
      }
      u |= zz[i];
    }
    uint32_t b = dzp_bits0(u);
    if ((size_t)(end - op) < 1 + (size_t)rem * wb + 16) return -1;
    *op++ = (uint8_t)b;
    op += dzp_pack(zz, rem, b, op);
  }
  return (int64_t)(op - out);
}

/* one 128-value PFoR frame (header + optional exception section + data) */
static uint8_t *dzp_frame_enc(const uint64_t *zz, uint32_t wbits,
                              uint8_t *op) {
  uint32_t b, bx;
  dzp_fit(zz, DZP_BLOCK, wbits, &b, &bx);
  if (bx == 0) {
    *op++ = (uint8_t)b;
    op += dzp_pack(zz, DZP_BLOCK, b, op);
  } else {
    *op++ = (uint8_t)(0x80 | b);
    *op++ = (uint8_t)bx;
    uint64_t msk = b >= 64 ? ~0ull : ((1ull << b) - 1);
    uint64_t lowv[DZP_BLOCK], exc[DZP_BLOCK];
    uint64_t xmap[DZP_BLOCK / 64] = {0, 0};
    uint32_t xn = 0;
    for (uint32_t i = 0; i < DZP_BLOCK; i++) {
      lowv[i] = zz[i] & msk;
      if (zz[i] > msk) {
        xmap[i >> 6] |= 1ull << (i & 63);
        exc[xn++] = zz[i] >> b;
      }
    }
    memcpy(op, xmap, DZP_BLOCK / 8);
    op += DZP_BLOCK / 8;
    op += dzp_pack(exc, xn, bx, op);
    op += dzp_pack(lowv, DZP_BLOCK, b, op);
  }
  return op;
}

static const uint8_t *dzp_frame_dec(const uint8_t *ip, const uint8_t *end,
                                    uint64_t *zz) {
  if (ip >= end) return NULL;
  uint8_t h = *ip++;
  uint32_t b = h & 0x7F, bx = 0;
  if (h & 0x80) {
    if (ip >= end) return NULL;
    bx = *ip++;
  }
  if (bx == 0) {
    if (!dzp_room(ip, end, DZP_BLOCK, b)) return NULL;
    dzp_unpack(ip, DZP_BLOCK, b, zz);
    ip += ((size_t)DZP_BLOCK * b + 7) / 8;
  } else {
    uint64_t xmap[2], exc[DZP_BLOCK];
    if (ip + 16 > end) return NULL;
    memcpy(xmap, ip, 16);
    ip += 16;
    uint32_t xn = (uint32_t)(__builtin_popcountll(xmap[0]) +
                             __builtin_popcountll(xmap[1]));
    if (!dzp_room(ip, end, xn, bx)) return NULL;
    dzp_unpack(ip, xn, bx, exc);
    ip += ((size_t)xn * bx + 7) / 8;
    if (!dzp_room(ip, end, DZP_BLOCK, b)) return NULL;
    dzp_unpack(ip, DZP_BLOCK, b, zz);
    ip += ((size_t)DZP_BLOCK * b + 7) / 8;
    uint32_t xi = 0;
    for (uint32_t i = 0; i < DZP_BLOCK; i++)
      if ((xmap[i >> 6] >> (i & 63)) & 1) zz[i] |= exc[xi++] << b;
  }
  return ip;
}

int64_t obx_cs_dzp_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                       uint8_t *out, size_t cap) {
  return dzp_enc_core(in, count, wb, out, cap, 1);
}

/* XOR_FIXED_PFOR (ObXorFixedPforInner, ob_xor_fixed_pfor.h:22-175):
 * per 128 block: v = cur ^ prev; shift = wbits - bits(OR of xors); each
 * value is (v << shift) BIT-REVERSED at element width; frame =
 * [shift byte][PFoR frame of the reversed values]. Tail:
 * [shift byte][SimpleBitPacking (own maxbits byte)]. */
static inline uint64_t dzp_bitrev(uint64_t v, uint32_t wbits) {
  uint64_t r = 0;
  for (uint32_t i = 0; i < wbits; i++) {
    r = (r << 1) | (v & 1);
    v >>= 1;
  }
  return r;
}

int64_t obx_cs_xpfor_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                         uint8_t *out, size_t cap) {
  uint32_t wbits = wb * 8;
  uint64_t wmask = wbits >= 64 ? ~0ull : ((1ull << wbits) - 1);
  uint8_t *op = out;
  uint8_t *end = out + cap;
  uint64_t start = 0;
  uint64_t v[DZP_BLOCK];
  uint32_t done = 0;
  while (count - done >= DZP_BLOCK) {
    uint64_t orr = 0;
    for (uint32_t i = 0; i < DZP_BLOCK; i++) {
      uint64_t cur = dzr_load(in + (size_t)(done + i) * wb, wb);
      v[i] = (cur ^ start) & wmask;
      orr |= v[i];
      start = cur;
    }
    uint32_t sh = wbits - dzp_bits0(orr);
    for (uint32_t i = 0; i < DZP_BLOCK; i++)
      v[i] = dzp_bitrev((v[i] << sh) & wmask, wbits);
    if ((size_t)(end - op) < 2 + 16 + (size_t)DZP_BLOCK * wb + 32)
      return -1;
    *op++ = (uint8_t)sh;
    op = dzp_frame_enc(v, wbits, op);
    done += DZP_BLOCK;
  }
  if (done < count) {
    uint32_t rem = count - done;
    uint64_t orr = 0;
    for (uint32_t i = 0; i < rem; i++) {
      uint64_t cur = dzr_load(in + (size_t)(done + i) * wb, wb);
      v[i] = (cur ^ start) & wmask;
      orr |= v[i];
      start = cur;
    }
    uint32_t sh = wbits - dzp_bits0(orr);
    uint64_t u = 0;
    for (uint32_t i = 0; i < rem; i++) {
      v[i] = dzp_bitrev((v[i] << sh) & wmask, wbits);
      u |= v[i];
    }
    uint32_t b = dzp_bits0(u);
    if ((size_t)(end - op) < 2 + (size_t)rem * wb + 16) return -1;
    *op++ = (uint8_t)sh;
    *op++ = (uint8_t)b; /* SimpleBitPacking maxbits byte */
    op += dzp_pack(v, rem, b, op);
  }
  return (int64_t)(op - out);
}

int64_t obx_cs_xpfor_dec(const uint8_t *in, size_t in_len, uint32_t count,
                         uint32_t wb, uint8_t *out) {
  uint32_t wbits = wb * 8;
  uint64_t wmask = wbits >= 64 ? ~0ull : ((1ull << wbits) - 1);
  const uint8_t *ip = in;
  const uint8_t *end = in + in_len;
  uint64_t start = 0;
  uint64_t v[DZP_BLOCK];
  uint32_t done = 0;
  while (count - done >= DZP_BLOCK) {
    if (ip >= end) return -1;
    uint32_t sh = *ip++;
    ip = dzp_frame_dec(ip, end, v);
    if (!ip) return -1;
    for (uint32_t i = 0; i < DZP_BLOCK; i++) {
      uint64_t x = (dzp_bitrev(v[i] & wmask, wbits) >> sh) & wmask;
      start = (x ^ start) & wmask;
      memcpy(out + (size_t)(done + i) * wb, &start, wb);
    }
    done += DZP_BLOCK;
  }
  if (done < count) {
    uint32_t rem = count - done;
    if (ip + 2 > end) return -1;
    uint32_t sh = *ip++;
    uint32_t b = *ip++;
    if (!dzp_room(ip, end, rem, b)) return -1;
    dzp_unpack(ip, rem, b, v);
    ip += ((size_t)rem * b + 7) / 8;
    for (uint32_t i = 0; i < rem; i++) {
      uint64_t x = (dzp_bitrev(v[i] & wmask, wbits) >> sh) & wmask;
      start = (x ^ start) & wmask;
      memcpy(out + (size_t)(done + i) * wb, &start, wb);
    }
  }
  return (int64_t)(ip - in);
}

int64_t obx_cs_ddzp_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                        uint8_t *out, size_t cap) {
  return dzp_enc_core(in, count, wb, out, cap, 2);
}

int64_t obx_cs_fpfor_enc(const uint8_t *in, uint32_t count, uint32_t wb,
                         uint8_t *out, size_t cap) {
  return dzp_enc_core(in, count, wb, out, cap, 0);
}

static int64_t dzp_dec_core(const uint8_t *in, size_t in_len,
                            uint32_t count, uint32_t wb, uint8_t *out,
                            int transform) {
  uint32_t wbits = wb * 8;
  uint64_t wmask = wbits >= 64 ? ~0ull : ((1ull << wbits) - 1);
  const uint8_t *ip = in;
  const uint8_t *end = in + in_len;
  uint64_t start = 0, pd = 0;
  uint64_t zz[DZP_BLOCK], exc[DZP_BLOCK];
  uint32_t done = 0;
  while (count - done >= DZP_BLOCK) {
    if (ip >= end) return -1;
    uint8_t h = *ip++;
    /* 7-bit width field; the VByte variant is commented out in the
       reference encoder (ob_simd_fixed_pfor.h:177) */
    uint32_t b = h & 0x7F, bx = 0;
    if (h & 0x80) {
      if (ip >= end) return -1;
      bx = *ip++;
    }
    if (bx == 0) {
      if (!dzp_room(ip, end, DZP_BLOCK, b)) return -1;
      dzp_unpack(ip, DZP_BLOCK, b, zz);
      ip += ((size_t)DZP_BLOCK * b + 7) / 8;
    } else {
      uint64_t xmap[2];
      if (ip + 16 > end) return -1;
      memcpy(xmap, ip, 16);
      ip += 16;
      uint32_t xn = (uint32_t)(__builtin_popcountll(xmap[0]) +
                               __builtin_popcountll(xmap[1]));
      if (!dzp_room(ip, end, xn, bx)) return -1;
      dzp_unpack(ip, xn, bx, exc);
      ip += ((size_t)xn * bx + 7) / 8;
      if (!dzp_room(ip, end, DZP_BLOCK, b)) return -1;
      dzp_unpack(ip, DZP_BLOCK, b, zz);
      ip += ((size_t)DZP_BLOCK * b + 7) / 8;
      uint32_t xi = 0;
      for (uint32_t i = 0; i < DZP_BLOCK; i++)
        if ((xmap[i >> 6] >> (i & 63)) & 1) zz[i] |= exc[xi++] << b;
    }
    for (uint32_t i = 0; i < DZP_BLOCK; i++) {
      if (transform == 0) {
        start = zz[i] & wmask;
      } else if (transform == 1) {
        start = (start + dzr_zz_dec(zz[i], wbits)) & wmask;
      } else {
        pd = (pd + dzr_zz_dec(zz[i], wbits)) & wmask;
        start = (start + pd) & wmask;
      }
      memcpy(out + (size_t)(done + i) * wb, &start, wb);
    }
    done += DZP_BLOCK;
  }
  if (done < count) {
    uint32_t rem = count - done;
    if (ip >= end) return -1;
    uint32_t b = *ip++;
    if (!dzp_room(ip, end, rem, b)) return -1;
    dzp_unpack(ip, rem, b, zz);
    ip += ((size_t)rem * b + 7) / 8;
    for (uint32_t i = 0; i < rem; i++) {
      if (transform == 0) {
        start = zz[i] & wmask;
      } else if (transform == 1) {
        start = (start + dzr_zz_dec(zz[i], wbits)) & wmask;
      } else {
        pd = (pd + dzr_zz_dec(zz[i], wbits)) & wmask;
        start = (start + pd) & wmask;
      }
      memcpy(out + (size_t)(done + i) * wb, &start, wb);
    }
  }
  return (int64_t)(ip - in);
}

int64_t obx_cs_dzp_dec(const uint8_t *in, size_t in_len, uint32_t count,
                       uint32_t wb, uint8_t *out) {
  return dzp_dec_core(in, in_len, count, wb, out, 1);
}

int64_t obx_cs_ddzp_dec(const uint8_t *in, size_t in_len, uint32_t count,
                        uint32_t wb, uint8_t *out) {
  return dzp_dec_core(in, in_len, count, wb, out, 2);
}

int64_t obx_cs_fpfor_dec(const uint8_t *in, size_t in_len, uint32_t count,
                         uint32_t wb, uint8_t *out) {
  return dzp_dec_core(in, in_len, count, wb, out, 0);
}
