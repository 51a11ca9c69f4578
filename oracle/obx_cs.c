/* CS integer-stream restatement — see obx_cs.h for the cited reference
 * functions. Oracle/test infrastructure only. */
#include "obx_cs.h"

#include <string.h>

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
  if (!vals || !rows || !buf) return -1;
  /* build_signed_stream_meta: base = min, width covers range = max-min */
  int64_t mn = 0, mx = 0;
  int any = 0;
  for (uint32_t r = 0; r < rows; r++) {
    if (null_at(nulls, r)) continue;
    if (!any || vals[r] < mn) mn = vals[r];
    if (!any || vals[r] > mx) mx = vals[r];
    any = 1;
  }
  if (!any) mn = mx = 0;
  uint64_t range = (uint64_t)mx - (uint64_t)mn;
  obx_cs_int_meta m;
  memset(&m, 0, sizeof(m));
  m.version = 1; /* V2 */
  m.type = OBX_CS_ENC_RAW;
  m.attr = OBX_CS_USE_BASE;
  m.base = (uint64_t)mn;
  m.width_tag = range <= 0xFF ? 0 : range <= 0xFFFF ? 1
                : range <= 0xFFFFFFFFull ? 2 : 3;
  m.pfor_packing_type = 0; /* CPU_ARCH_INDEPENDANT_SCALAR */
  int hn = obx_cs_int_meta_enc(&m, buf, cap);
  if (hn < 0) return -1;
  size_t pos = (size_t)hn;
  uint32_t wb = WIDTH_BYTES[m.width_tag];
  if (pos + (size_t)rows * wb > cap) return -1;
  for (uint32_t r = 0; r < rows; r++) {
    /* null -> replace value (== base here: "make int small",
       ob_integer_stream_encoder.cpp:108-112), base subtracted FIRST */
    uint64_t ele = null_at(nulls, r) ? m.base : (uint64_t)vals[r];
    ele -= m.base;
    memcpy(buf + pos, &ele, wb);
    pos += wb;
  }
  return (int64_t)pos;
}

int64_t obx_cs_int_stream_dec(const uint8_t *buf, size_t len, uint32_t rows,
                              int64_t *out, obx_cs_int_meta *meta_out) {
  obx_cs_int_meta m;
  int hn = obx_cs_int_meta_dec(buf, len, &m);
  if (hn < 0 || m.type != OBX_CS_ENC_RAW) return -1;
  size_t pos = (size_t)hn;
  uint32_t wb = WIDTH_BYTES[m.width_tag];
  if (pos + (size_t)rows * wb > len) return -1;
  uint64_t base = (m.attr & OBX_CS_USE_BASE) ? m.base : 0;
  for (uint32_t r = 0; r < rows; r++) {
    uint64_t ele = 0;
    memcpy(&ele, buf + pos, wb);
    pos += wb;
    out[r] = (int64_t)(ele + base);
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
