/*
 * obx_cs_load.cpp — product-side CS (cs_encoding) block loading: the
 * load-time transform of ObCSMicroBlockTransformer
 * (/root/reference/src/storage/blocksstable/cs_encoding/
 * ob_cs_micro_block_transformer.cpp), GPU-native.
 *
 * The host parses CS block/stream METADATA only (headers, stream slices,
 * integer/string stream metas — the cached-decoder position of
 * ObMicroBlockCSDecoder); all value decoding (RAW widths, the PFoR and
 * RLE codec families, dict refs, null recovery) runs on the GPU
 * (obx_cs_kernels.hip) straight into an HBM arena laid out in the
 * engine's native scan form, which the ordinary PAX scan kernels then
 * read. liboracle.so is NEVER touched: this parser is an independent
 * restatement of the format (cites below), and parity tests compare the
 * loaded handle's decode output against the CPU oracle bit-exactly.
 *
 * Format cites:
 *   ob_column_encoding_struct.h:24-169   ObCSColumnHeader/ObAllColumnHeader
 *   ob_micro_block_cs_encoder.cpp:1387   block layout [hdr][ach][col hdrs]
 *                                        [col data][string pool][offsets]
 *   ob_stream_encoding_struct.{h,cpp}    ObIntegerStreamMeta /
 *                                        ObStringStreamMeta serialize
 *   ob_integer_column_encoder.cpp:100    null replace-value rule
 *   ob_dict_column_encoder.cpp           dict meta, CONST_ENCODING_REF
 *   serialization.h:297                  vi64/vi32 varints
 *
 * Container choices shared with the writer (documented in
 * oracle/obx_cs_block.h): the 16-byte obx CS block header; the block-tail
 * stream-offset stream and dict ref streams are RAW-encoded; the string
 * pool is uncompressed.
 */
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstring>
#include <vector>

#include "obx_cs_dev.h"
#include "obx_dev.h"

#ifndef OBX_WG_HOST
#define OBX_WG_HOST 256
#endif
#include "../../include/obx.h"
#include "../../oracle/obx_format.h"

extern "C" __global__ void k_cs_decode(const uint8_t *, const cs_dev_stream *,
                                       uint32_t, uint8_t *);

/* from obx_engine.cpp */
struct obx_gpu_ctx;
struct obx_handle;
int obx_cs_finish_load(obx_gpu_ctx *ctx, obx_handle *h,
                       const obx_col_schema *cols, uint16_t n_cols,
                       std::vector<dev_block> &blocks, uint8_t *d_buf,
                       uint64_t arena_bytes, uint64_t total_rows);
int obx_cs_alloc_handle(obx_gpu_ctx *ctx, obx_handle **h_out);
int obx_cs_publish_handle(obx_gpu_ctx *ctx, obx_handle *h);

namespace {

/* ---- varints / metas (host parse) -------------------------------------- */
int vi64_dec(const uint8_t *p, size_t len, int64_t *out) {
  uint64_t v = 0;
  int sh = 0;
  for (int i = 0; i < 10 && (size_t)i < len; i++) {
    v |= (uint64_t)(p[i] & 0x7F) << sh;
    if (!(p[i] & 0x80)) { *out = (int64_t)v; return i + 1; }
    sh += 7;
  }
  return -1;
}
int vi32_dec(const uint8_t *p, size_t len, int32_t *out) {
  uint32_t v = 0;
  int sh = 0;
  for (int i = 0; i < 5 && (size_t)i < len; i++) {
    v |= (uint32_t)(p[i] & 0x7F) << sh;
    if (!(p[i] & 0x80)) { *out = (int32_t)v; return i + 1; }
    sh += 7;
  }
  return -1;
}

struct int_meta {
  uint8_t version, attr, type, width_tag;
  uint64_t base = 0;
  uint64_t null_replaced = 0;
  uint8_t precision_tag = 0, pfor_type = 0;
};
/* ObIntegerStream attr bits */
enum { CS_USE_BASE = 1, CS_REPLACE_NULL = 2, CS_DECIMAL_INT = 4 };

int int_meta_dec(const uint8_t *p, size_t len, int_meta *m) {
  if (len < 4) return -1;
  size_t pos = 0;
  m->version = p[pos++];
  m->attr = p[pos++];
  m->type = p[pos++];
  m->width_tag = p[pos++];
  if (m->attr & CS_USE_BASE) {
    int64_t v;
    int n = vi64_dec(p + pos, len - pos, &v);
    if (n < 0) return -1;
    m->base = (uint64_t)v;
    pos += n;
  }
  if (m->attr & CS_REPLACE_NULL) {
    int64_t v;
    int n = vi64_dec(p + pos, len - pos, &v);
    if (n < 0) return -1;
    m->null_replaced = (uint64_t)v;
    pos += n;
  }
  if (m->attr & CS_DECIMAL_INT) {
    if (pos >= len) return -1;
    m->precision_tag = p[pos++];
  }
  if (m->version >= 1) {
    if (pos >= len) return -1;
    m->pfor_type = p[pos++];
  }
  return (int)pos;
}

struct str_meta {
  uint8_t version, attr;
  uint32_t uncompressed_len = 0;
  uint32_t fixed_str_len = 0;
};
enum { CS_STR_ZERO_LEN_NULL = 1, CS_STR_FIXED_LEN = 2, CS_STR_USE_ZERO_LEN = 4 };

int str_meta_dec(const uint8_t *p, size_t len, str_meta *m) {
  if (len < 2) return -1;
  size_t pos = 0;
  m->version = p[pos++];
  m->attr = p[pos++];
  int32_t v;
  int n = vi32_dec(p + pos, len - pos, &v);
  if (n < 0) return -1;
  m->uncompressed_len = (uint32_t)v;
  pos += n;
  if (m->attr & CS_STR_FIXED_LEN) {
    n = vi32_dec(p + pos, len - pos, &v);
    if (n < 0) return -1;
    m->fixed_str_len = (uint32_t)v;
    pos += n;
  }
  return (int)pos;
}

const uint32_t WB[4] = {1, 2, 4, 8};

/* decode a RAW integer stream host-side (used only for the block-tail
 * stream-offset stream and ref_row_cnt-sized metadata; the container
 * writes those RAW — obx_cs_block.h "deviations") */
int host_int_stream_dec(const uint8_t *p, size_t len, uint32_t count,
                        std::vector<uint64_t> &out) {
  int_meta m;
  int hn = int_meta_dec(p, len, &m);
  if (hn < 0 || m.type != 1 /* RAW */) return -1;
  uint32_t wb = WB[m.width_tag & 3];
  if ((size_t)hn + (size_t)count * wb > len) return -1;
  uint64_t base = (m.attr & CS_USE_BASE) ? m.base : 0;
  out.resize(count);
  const uint8_t *d = p + hn;
  for (uint32_t i = 0; i < count; i++) {
    uint64_t v = 0;
    memcpy(&v, d + (size_t)i * wb, wb);
    out[i] = v + base;
  }
  return 0;
}

/* ---- CS block structures (mirrors oracle/obx_cs_block.h formats) ------- */
#pragma pack(push, 1)
struct cs_bhdr {
  uint32_t magic;
  uint16_t version, header_size;
  uint32_t row_count;
  uint16_t column_count, reserved;
};
struct cs_ach {
  uint8_t version, attrs;
  uint32_t all_string_data_length, stream_offsets_length;
  uint16_t stream_count;
};
struct cs_chdr { uint8_t version, type, attrs, obj_type; };
struct cs_dmeta {
  uint8_t version, attrs;
  uint32_t distinct_val_cnt, ref_row_cnt;
};
#pragma pack(pop)

enum { CT_INT = 0, CT_STR = 1, CT_IDICT = 2, CT_SDICT = 3 };
enum { CA_FIXED = 1, CA_NULL_BM = 2 };
enum { DA_SORTED = 1, DA_HAS_NULL = 2, DA_CONST_REF = 4 };
constexpr uint32_t CS_MAGIC = 0x5343424Fu;

struct col_parse {
  cs_chdr h;
  uint64_t nb_off = 0;        /* null bitmap (abs in src), 0 = none */
  uint64_t istream_off = 0;   /* int stream (abs), after nothing */
  uint32_t istream_len = 0;
  int_meta im;                /* parsed int stream meta (values/refs) */
  uint64_t idata_off = 0;     /* int stream packed data (abs) */
  uint32_t idata_len = 0;
  str_meta sm;
  uint64_t sbytes_off = 0;    /* string bytes in the pool (abs) */
  cs_dmeta dm;
  uint64_t rstream_off = 0;   /* dict ref stream (abs) */
  uint32_t rstream_len = 0;
  int_meta rm;                /* ref stream meta */
  uint64_t rdata_off = 0;
  uint32_t rdata_len = 0;
};

} // namespace

/* exported for tests: parse one CS block's metadata; returns 0 or neg */
extern "C" int obx_cs_host_parse(const uint8_t *buf, int64_t len,
                                 uint32_t *rows_out, uint32_t *ncols_out) {
  if (len < (int64_t)sizeof(cs_bhdr)) return OBX_INVALID_ARGUMENT;
  cs_bhdr bh;
  memcpy(&bh, buf, sizeof(bh));
  if (bh.magic != CS_MAGIC || bh.version != 1) return OBX_INVALID_ARGUMENT;
  if (rows_out) *rows_out = bh.row_count;
  if (ncols_out) *ncols_out = bh.column_count;
  return OBX_SUCCESS;
}

static int parse_cs_block(const uint8_t *buf, size_t len, uint64_t src_base,
                          uint32_t *rows_out,
                          std::vector<col_parse> &cols_out) {
  cs_bhdr bh;
  if (len < sizeof(bh)) return OBX_INVALID_ARGUMENT;
  memcpy(&bh, buf, sizeof(bh));
  if (bh.magic != CS_MAGIC || bh.version != 1 || bh.column_count == 0 ||
      bh.column_count > 48)
    return OBX_INVALID_ARGUMENT;
  size_t hp = bh.header_size;
  cs_ach ach;
  if (hp + sizeof(ach) > len) return OBX_INVALID_ARGUMENT;
  memcpy(&ach, buf + hp, sizeof(ach));
  hp += sizeof(ach);
  if (hp + (size_t)bh.column_count * sizeof(cs_chdr) > len)
    return OBX_INVALID_ARGUMENT;
  cols_out.resize(bh.column_count);
  for (uint32_t c = 0; c < bh.column_count; c++)
    memcpy(&cols_out[c].h, buf + hp + c * sizeof(cs_chdr), sizeof(cs_chdr));
  hp += (size_t)bh.column_count * sizeof(cs_chdr);

  if ((size_t)ach.stream_offsets_length + ach.all_string_data_length > len)
    return OBX_INVALID_ARGUMENT;
  const size_t so_start = len - ach.stream_offsets_length;
  const size_t pool = so_start - ach.all_string_data_length;
  std::vector<uint64_t> ends;
  if (ach.stream_count) {
    if (host_int_stream_dec(buf + so_start, ach.stream_offsets_length,
                            ach.stream_count, ends) < 0)
      return OBX_INVALID_ARGUMENT;
    for (auto e : ends)
      if (e > pool) return OBX_INVALID_ARGUMENT;
  }

  const uint32_t rows = bh.row_count;
  const uint32_t bm = (rows + 7) / 8;
  size_t pos = hp;
  uint32_t si = 0, str_off = 0;
  auto next_stream = [&](uint64_t *off, uint32_t *slen) -> int {
    if (si >= ends.size() || ends[si] < pos) return -1;
    *off = src_base + pos;
    *slen = (uint32_t)(ends[si] - pos);
    pos = ends[si++];
    return 0;
  };
  for (uint32_t c = 0; c < bh.column_count; c++) {
    col_parse &cp = cols_out[c];
    if (cp.h.attrs & CA_NULL_BM) {
      if (pos + bm > len) return OBX_INVALID_ARGUMENT;
      cp.nb_off = src_base + pos;
      pos += bm;
    }
    auto parse_imeta = [&](uint64_t off, uint32_t slen, int_meta *m,
                           uint64_t *doff, uint32_t *dlen) -> int {
      int hn = int_meta_dec(buf + (off - src_base), slen, m);
      if (hn < 0) return -1;
      *doff = off + hn;
      *dlen = slen - hn;
      return 0;
    };
    switch (cp.h.type) {
      case CT_INT: {
        if (next_stream(&cp.istream_off, &cp.istream_len) < 0 ||
            parse_imeta(cp.istream_off, cp.istream_len, &cp.im,
                        &cp.idata_off, &cp.idata_len) < 0)
          return OBX_INVALID_ARGUMENT;
        break;
      }
      case CT_STR: {
        uint64_t off; uint32_t slen;
        if (next_stream(&off, &slen) < 0) return OBX_INVALID_ARGUMENT;
        if (str_meta_dec(buf + (off - src_base), slen, &cp.sm) < 0)
          return OBX_INVALID_ARGUMENT;
        cp.sbytes_off = src_base + pool + str_off;
        str_off += cp.sm.uncompressed_len;
        if (str_off > ach.all_string_data_length)
          return OBX_INVALID_ARGUMENT;
        if (!(cp.sm.attr & CS_STR_FIXED_LEN)) {
          /* var-length strings: no fixed-cell scan form */
          return OBX_NOT_SUPPORTED;
        }
        break;
      }
      case CT_IDICT:
      case CT_SDICT: {
        if (pos + sizeof(cs_dmeta) > len) return OBX_INVALID_ARGUMENT;
        memcpy(&cp.dm, buf + pos, sizeof(cs_dmeta));
        pos += sizeof(cs_dmeta);
        if (cp.dm.distinct_val_cnt == 0) break; /* all-null column */
        if (cp.h.type == CT_IDICT) {
          if (next_stream(&cp.istream_off, &cp.istream_len) < 0 ||
              parse_imeta(cp.istream_off, cp.istream_len, &cp.im,
                          &cp.idata_off, &cp.idata_len) < 0)
            return OBX_INVALID_ARGUMENT;
        } else {
          uint64_t off; uint32_t slen;
          if (next_stream(&off, &slen) < 0) return OBX_INVALID_ARGUMENT;
          if (str_meta_dec(buf + (off - src_base), slen, &cp.sm) < 0)
            return OBX_INVALID_ARGUMENT;
          cp.sbytes_off = src_base + pool + str_off;
          str_off += cp.sm.uncompressed_len;
          if (str_off > ach.all_string_data_length)
            return OBX_INVALID_ARGUMENT;
          if (!(cp.sm.attr & CS_STR_FIXED_LEN)) return OBX_NOT_SUPPORTED;
        }
        if (next_stream(&cp.rstream_off, &cp.rstream_len) < 0 ||
            parse_imeta(cp.rstream_off, cp.rstream_len, &cp.rm,
                        &cp.rdata_off, &cp.rdata_len) < 0)
          return OBX_INVALID_ARGUMENT;
        break;
      }
      default:
        return OBX_NOT_SUPPORTED;
    }
  }
  *rows_out = rows;
  return OBX_SUCCESS;
}

/* ---- arena planning + the load entry point ----------------------------- */

extern "C" int obx_gpu_load_cs_blocks(obx_gpu_ctx *ctx,
                                      const obx_blockset *bs) {
  if (!ctx || !bs || !bs->data || !bs->cols) return OBX_INVALID_ARGUMENT;
  const uint16_t n_cols = bs->n_cols;
  if (n_cols == 0 || n_cols > OBX_DEV_MAX_COLS) return OBX_NOT_SUPPORTED;
  for (uint16_t c = 0; c < n_cols; c++) {
    int sc = obx_store_class(bs->cols[c].obj_type);
    if (sc == OBX_SC_STRING) {
      if (bs->cols[c].len == 0 || bs->cols[c].len > 8)
        return OBX_NOT_SUPPORTED; /* engine cells are <= 8 bytes */
    } else if (bs->cols[c].len != 8 && bs->cols[c].len != 4) {
      return OBX_NOT_SUPPORTED;
    }
  }

  /* pass 1: parse every block, plan arena + scratch + stream tasks */
  struct chunk_plan {
    dev_block db;
  };
  std::vector<dev_block> blocks;
  std::vector<cs_dev_stream> p1, p2;
  uint64_t arena = 0;   /* chunk region cursor */
  uint64_t total_rows = 0;
  const uint64_t src_total = bs->block_offsets[bs->n_blocks];

  /* scratch tail starts after all chunks — two-pass: first compute chunk
     bytes, then assign scratch; simpler: collect tasks with scratch
     offsets relative to a scratch cursor, fix up after */
  uint64_t scratch = 0; /* relative cursor; rebased later */
  struct fixup { size_t idx; int which; }; /* which: 1 src_byte, 2 out_byte */
  std::vector<fixup> fx1, fx2;
  auto scratch_alloc = [&](uint64_t bytes) {
    uint64_t off = scratch;
    scratch += (bytes + 15) & ~15ull;
    return off;
  };

  for (uint32_t b = 0; b < bs->n_blocks; b++) {
    const uint8_t *blk = bs->data + bs->block_offsets[b];
    size_t blen = bs->block_offsets[b + 1] - bs->block_offsets[b];
    uint32_t rows = 0;
    std::vector<col_parse> cp;
    int rc = parse_cs_block(blk, blen, bs->block_offsets[b], &rows, cp);
    if (rc != OBX_SUCCESS) return rc;
    if (cp.size() != n_cols) return OBX_INVALID_ARGUMENT;

    /* per-block scratch: decoded int64 values / entries / refs */
    std::vector<uint64_t> sc_vals(n_cols, ~0ull), sc_ent(n_cols, ~0ull),
        sc_refs(n_cols, ~0ull);
    for (uint16_t c = 0; c < n_cols; c++) {
      const col_parse &p = cp[c];
      if (p.h.type == CT_INT) {
        sc_vals[c] = scratch_alloc((uint64_t)rows * 8);
        cs_dev_stream s{};
        s.src_byte = p.idata_off;
        s.src_len = p.idata_len;
        s.count = rows;
        s.base = (p.im.attr & CS_USE_BASE) ? (int64_t)p.im.base : 0;
        s.op = OBX_CSOP_INT64;
        s.enc_type = p.im.type;
        s.wb = (uint8_t)WB[p.im.width_tag & 3];
        fx2.push_back({p1.size(), 2});
        s.out_byte = sc_vals[c];
        p1.push_back(s);
      } else if ((p.h.type == CT_IDICT || p.h.type == CT_SDICT) &&
                 p.dm.distinct_val_cnt > 0) {
        if (p.h.type == CT_IDICT) {
          sc_ent[c] = scratch_alloc((uint64_t)p.dm.distinct_val_cnt * 8);
          cs_dev_stream s{};
          s.src_byte = p.idata_off;
          s.src_len = p.idata_len;
          s.count = p.dm.distinct_val_cnt;
          s.base = (p.im.attr & CS_USE_BASE) ? (int64_t)p.im.base : 0;
          s.op = OBX_CSOP_INT64;
          s.enc_type = p.im.type;
          s.wb = (uint8_t)WB[p.im.width_tag & 3];
          fx2.push_back({p1.size(), 2});
          s.out_byte = sc_ent[c];
          p1.push_back(s);
        }
        uint32_t rcnt = p.dm.ref_row_cnt;
        sc_refs[c] = scratch_alloc((uint64_t)rcnt * 8);
        cs_dev_stream s{};
        s.src_byte = p.rdata_off;
        s.src_len = p.rdata_len;
        s.count = rcnt;
        s.base = (p.rm.attr & CS_USE_BASE) ? (int64_t)p.rm.base : 0;
        s.op = OBX_CSOP_INT64;
        s.enc_type = p.rm.type;
        s.wb = (uint8_t)WB[p.rm.width_tag & 3];
        fx2.push_back({p1.size(), 2});
        s.out_byte = sc_refs[c];
        p1.push_back(s);
      }
    }

    /* chunking: rows per chunk so the chunk fits the LDS stage and the
       per-chunk row cap; rows multiple of 8 keeps bitmaps byte-aligned */
    auto chunk_bytes = [&](uint32_t n) -> uint64_t {
      uint64_t t = 0;
      for (uint16_t c = 0; c < n_cols; c++) {
        const col_parse &p = cp[c];
        bool nulls = p.nb_off || (p.im.attr & CS_REPLACE_NULL);
        switch (p.h.type) {
          case CT_INT:
            t += (nulls ? (n + 7) / 8 : 0);
            t = (t + 7) & ~7ull;
            t += (uint64_t)n * 8;
            break;
          case CT_STR:
            t += (p.nb_off ? (n + 7) / 8 : 0);
            t += (uint64_t)n * bs->cols[c].len;
            break;
          default: { /* dict or all-null */
            if (p.dm.distinct_val_cnt == 0) break; /* CONST null */
            uint32_t el = p.h.type == CT_IDICT ? 8 : p.sm.fixed_str_len;
            uint32_t rw = p.dm.distinct_val_cnt > 255 ? 2 : 1;
            t += (uint64_t)p.dm.distinct_val_cnt * el;
            t += (uint64_t)n * rw;
            break;
          }
        }
        t = (t + 7) & ~7ull;
      }
      return t;
    };
    uint32_t crows = rows < 2048 ? rows : 2048;
    while (crows > 8 && chunk_bytes(crows) > OBX_LDS_STAGE_BYTES - 64)
      crows = (crows - 8) & ~7u;
    if (chunk_bytes(crows) > OBX_LDS_STAGE_BYTES - 64)
      return OBX_NOT_SUPPORTED; /* pathological wide schema */

    for (uint32_t r0 = 0; r0 < rows; r0 += crows) {
      const uint32_t n = (rows - r0) < crows ? (rows - r0) : crows;
      const uint64_t cbase = arena; /* 16-aligned below */
      uint64_t off = 0;             /* within chunk */
      dev_block db{};
      db.row_start_lo = (uint32_t)(total_rows + r0);
      db.row_start_hi = (uint32_t)((total_rows + r0) >> 32);
      db.row_count = n;
      db.block_byte = cbase;
      for (uint16_t c = 0; c < n_cols; c++) {
        const col_parse &p = cp[c];
        dev_col *dc = &db.cols[c];
        const int sc2 = obx_store_class(bs->cols[c].obj_type);
        dc->datum_len = bs->cols[c].len;
        dc->tss = 8;
        if (sc2 == OBX_SC_INT) dc->flags |= OBX_DF_SIGNED;
        if (sc2 == OBX_SC_STRING) dc->flags |= OBX_DF_STRING;
        switch (p.h.type) {
          case CT_INT: {
            bool nulls = p.nb_off || (p.im.attr & CS_REPLACE_NULL);
            dc->enc = OBX_D_RAW;
            dc->width = 8;
            dc->ext_width = 1;
            if (nulls) {
              dc->flags |= OBX_DF_HAS_EXT;
              dc->ext_bit = (cbase + off) * 8;
              cs_dev_stream s{};
              s.count = n;
              s.out_byte = cbase + off;
              if (p.im.attr & CS_REPLACE_NULL) {
                s.op = OBX_CSOP_EXT_REPLACE;
                s.base = (int64_t)p.im.null_replaced;
                s.src_byte = sc_vals[c] + (uint64_t)r0 * 8;
                fx1.push_back({p2.size(), 1});
                p2.push_back(s);
              } else {
                s.op = OBX_CSOP_EXT_BITMAP;
                s.src_byte = p.nb_off + r0 / 8;
                p1.push_back(s);
              }
              off += (n + 7) / 8;
            }
            off = (off + 7) & ~7ull;
            dc->data_bit = (cbase + off) * 8;
            {
              cs_dev_stream s{};
              s.op = OBX_CSOP_I64_TO_BYTES;
              s.count = n;
              s.wb = 8;
              s.src_byte = sc_vals[c] + (uint64_t)r0 * 8;
              s.out_byte = cbase + off;
              fx1.push_back({p2.size(), 1});
              p2.push_back(s);
            }
            off += (uint64_t)n * 8;
            break;
          }
          case CT_STR: {
            dc->enc = OBX_D_RAW;
            dc->width = bs->cols[c].len;
            dc->ext_width = 1;
            if (p.nb_off) {
              dc->flags |= OBX_DF_HAS_EXT;
              dc->ext_bit = (cbase + off) * 8;
              cs_dev_stream s{};
              s.op = OBX_CSOP_EXT_BITMAP;
              s.count = n;
              s.src_byte = p.nb_off + r0 / 8;
              s.out_byte = cbase + off;
              p1.push_back(s);
              off += (n + 7) / 8;
            }
            dc->data_bit = (cbase + off) * 8;
            {
              cs_dev_stream s{};
              s.op = OBX_CSOP_COPY;
              s.src_byte = p.sbytes_off + (uint64_t)r0 * bs->cols[c].len;
              s.src_len = n * bs->cols[c].len;
              s.out_byte = cbase + off;
              p1.push_back(s);
            }
            off += (uint64_t)n * bs->cols[c].len;
            break;
          }
          default: { /* dict */
            if (p.dm.distinct_val_cnt == 0) { /* all-null: CONST null */
              dc->enc = OBX_D_CONST;
              dc->runs = 0;
              dc->count = 0;
              break;
            }
            uint32_t cnt = p.dm.distinct_val_cnt;
            uint32_t el = p.h.type == CT_IDICT ? 8 : p.sm.fixed_str_len;
            uint32_t rw = cnt > 255 ? 2 : 1;
            dc->enc = OBX_D_DICT;
            dc->count = cnt;
            dc->entry_len = (uint8_t)el;
            dc->width = (uint8_t)rw;
            dc->dict_byte = cbase + off;
            if (p.h.type == CT_IDICT) {
              cs_dev_stream s{};
              s.op = OBX_CSOP_I64_TO_BYTES;
              s.count = cnt;
              s.wb = 8;
              s.src_byte = sc_ent[c];
              s.out_byte = cbase + off;
              fx1.push_back({p2.size(), 1});
              p2.push_back(s);
            } else {
              cs_dev_stream s{};
              s.op = OBX_CSOP_COPY;
              s.src_byte = p.sbytes_off;
              s.src_len = cnt * el;
              s.out_byte = cbase + off;
              p1.push_back(s);
            }
            off += (uint64_t)cnt * el;
            dc->data_bit = (cbase + off) * 8;
            {
              cs_dev_stream s{};
              s.count = n;
              s.wb = (uint8_t)rw;
              s.out_byte = cbase + off;
              s.row0 = r0;
              if (p.dm.attrs & DA_CONST_REF) {
                s.op = OBX_CSOP_CONSTREF;
                s.src_byte = sc_refs[c];
              } else {
                s.op = OBX_CSOP_REFS;
                s.src_byte = sc_refs[c] + (uint64_t)r0 * 8;
              }
              fx1.push_back({p2.size(), 1});
              p2.push_back(s);
            }
            off += (uint64_t)n * rw;
            break;
          }
        }
        off = (off + 7) & ~7ull;
      }
      db.block_len = (uint32_t)off;
      blocks.push_back(db);
      arena += (off + 15) & ~15ull;
    }
    total_rows += rows;
  }

  /* rebase scratch offsets to the arena tail */
  const uint64_t scratch_base = (arena + 63) & ~63ull;
  const uint64_t arena_total = scratch_base + scratch + 64;
  for (auto &f : fx1) p2[f.idx].src_byte += scratch_base;
  for (auto &f : fx2) p1[f.idx].out_byte += scratch_base;

  /* device: upload src + tasks, decode in two phases into the arena */
  obx_handle *h = nullptr;
  int rc = obx_cs_alloc_handle(ctx, &h);
  if (rc != OBX_SUCCESS) return rc;
  uint8_t *d_src = nullptr, *d_buf = nullptr;
  cs_dev_stream *d_ss = nullptr;
  auto fail = [&](int code) {
    (void)hipFree(d_src);
    (void)hipFree(d_ss);
    (void)hipFree(d_buf);
    return code;
  };
  if (hipMalloc(&d_buf, arena_total) != hipSuccess)
    return fail(OBX_INTERNAL_ERROR);
  if (hipMalloc(&d_src, src_total + 64) != hipSuccess)
    return fail(OBX_INTERNAL_ERROR);
  if (hipMemcpy(d_src, bs->data, src_total, hipMemcpyHostToDevice) !=
      hipSuccess)
    return fail(OBX_INTERNAL_ERROR);
  size_t n1 = p1.size(), n2 = p2.size();
  if (hipMalloc(&d_ss, (n1 + n2) * sizeof(cs_dev_stream)) != hipSuccess)
    return fail(OBX_INTERNAL_ERROR);
  if (n1)
    (void)hipMemcpy(d_ss, p1.data(), n1 * sizeof(cs_dev_stream),
                    hipMemcpyHostToDevice);
  if (n2)
    (void)hipMemcpy(d_ss + n1, p2.data(), n2 * sizeof(cs_dev_stream),
                    hipMemcpyHostToDevice);
  auto grid = [&](size_t n) {
    uint32_t waves = (uint32_t)((n + 3) / 4);
    return waves < 4096 ? (waves ? waves : 1) : 4096;
  };
  if (n1)
    hipLaunchKernelGGL(k_cs_decode, dim3(grid(n1)), dim3(256), 0, nullptr,
                       d_src, d_ss, (uint32_t)n1, d_buf);
  if (n2)
    hipLaunchKernelGGL(k_cs_decode, dim3(grid(n2)), dim3(256), 0, nullptr,
                       d_src, d_ss + n1, (uint32_t)n2, d_buf);
  if (hipDeviceSynchronize() != hipSuccess) return fail(OBX_INTERNAL_ERROR);
  (void)hipFree(d_src);
  (void)hipFree(d_ss);
  d_src = nullptr;
  d_ss = nullptr;

  rc = obx_cs_finish_load(ctx, h, bs->cols, n_cols, blocks, d_buf,
                          arena_total, total_rows);
  if (rc != OBX_SUCCESS) return fail(rc);
  return obx_cs_publish_handle(ctx, h);
}
