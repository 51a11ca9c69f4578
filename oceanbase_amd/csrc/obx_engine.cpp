/*
 * obx_engine.cpp — MI355X product engine host side (C-ABI of include/obx.h).
 *
 * Replaces the reference's scan driver around the microblock reader
 * (ObSSTableRowScanner / ObMicroBlockRowScanner / ObVectorStore,
 * /root/reference/src/storage/access/ob_sstable_row_scanner.cpp:557-598,
 * ob_vector_store.cpp:329-389): microblocks are staged to HBM once, block
 * headers are parsed ONCE into flat device descriptors (the reference caches
 * per-column decoders the same way, ob_micro_block_decoder.cpp:440-480), and
 * the per-query work is pure GPU kernels (obx_kernels.hip).
 *
 * This is the PRODUCT path: it fails loudly (OBX_NO_GPU) when no HIP device
 * is present — it never falls back to the CPU oracle.
 */
#include <hip/hip_runtime.h>

#include <cstdlib>
#include <cstring>
#include <vector>
#include <algorithm>

#include "obx_dev.h"

#ifndef OBX_WG_HOST
#define OBX_WG_HOST 256  /* must match the kernels' WG */
#endif
#include "../../include/obx.h"
#include "../../oracle/obx_format.h"

/* kernels (obx_kernels.hip) */
extern "C" __global__ void k_decode_lds(
    const uint8_t *, const dev_block *, uint32_t, uint32_t, uint32_t,
    uint8_t *, uint8_t *);
extern "C" __global__ void k_decode_multi(
    const uint8_t *, const dev_block *, uint32_t, const uint16_t *,
    const uint8_t *, uint32_t, uint8_t *const *, uint8_t *const *);
extern "C" __global__ void k_scan_agg_direct(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, gslot *, unsigned long long *);
extern "C" __global__ void k_scan_filter_agg(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, gslot *, unsigned long long *);
extern "C" __global__ void k_scan_filter_agg_lds(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, gslot *, unsigned long long *);
extern "C" __global__ void k_filter(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, uint64_t *, int32_t *, uint32_t *,
    unsigned long long *);
extern "C" __global__ void k_filter_lds(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, uint64_t *, int32_t *, uint32_t *,
    unsigned long long *);
/* _prog variants: compiled with the postfix AND/OR combine-program path
   (reference: ObPushdownFilterExecutor AND/OR trees); separate entry points
   so the AND-only fast path keeps its lean register budget. */
extern "C" __global__ void k_scan_filter_agg_prog(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, gslot *, unsigned long long *);
extern "C" __global__ void k_scan_filter_agg_prog_lds(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, gslot *, unsigned long long *);
extern "C" __global__ void k_filter_prog(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, uint64_t *, int32_t *, uint32_t *,
    unsigned long long *);
extern "C" __global__ void k_filter_prog_lds(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *,
    const blk_leaf *, const dev_plan_hdr, uint64_t *, int32_t *, uint32_t *,
    unsigned long long *);
extern "C" __global__ void k_decode(
    const uint8_t *, const dev_block *, uint32_t, uint32_t, uint32_t,
    uint8_t *, uint8_t *);
extern "C" __global__ void k_group_pass(
    const uint8_t *, const dev_block *, uint32_t, const dev_plan_hdr,
    const uint64_t *, uint8_t *, gslot *, unsigned long long *);
extern "C" __global__ void k_agg_pass(
    const uint8_t *, const dev_block *, uint32_t, const dev_plan_hdr,
    uint32_t, const uint64_t *, const uint8_t *, gslot *);
extern "C" __global__ void k_lower_leaves(
    const uint8_t *, const dev_block *, uint32_t, const dev_leaf *, uint32_t,
    uint32_t, const int64_t *, blk_leaf *);
extern "C" __global__ void k_col_minmax(
    const uint8_t *, const dev_block *, uint32_t, uint32_t, int64_t *);

#define HIP_TRY(x)                                        \
  do {                                                    \
    hipError_t _e = (x);                                  \
    if (_e != hipSuccess) {                               \
      fprintf(stderr, "obx: HIP error %s at %s:%d\n",     \
              hipGetErrorString(_e), __FILE__, __LINE__); \
      return OBX_INTERNAL_ERROR;                          \
    }                                                     \
  } while (0)

#include <cstdio>

struct obx_handle {
  uint8_t *d_buf = nullptr;
  dev_block *d_blocks = nullptr;
  uint32_t n_blocks = 0;
  uint16_t n_cols = 0;
  uint64_t total_rows = 0;
  uint64_t total_bytes = 0;
  obx_col_schema cols[OBX_DEV_MAX_COLS];
  /* query scratch */
  blk_leaf *d_bleaves = nullptr;
  uint32_t bleaves_cap = 0; /* in (block,leaf) entries */
  dev_leaf *d_pleaves = nullptr;
  gslot *d_gtable = nullptr;
  unsigned long long *d_counters = nullptr;
  uint64_t *d_bitmap = nullptr;
  int32_t *d_row_ids = nullptr;
  uint32_t *d_blk_counts = nullptr;
  uint8_t *d_row_slot = nullptr;
  uint8_t *d_decode_out[OBX_DEV_MAX_COLS] = {};
  uint64_t last_survivors = 0;
  gslot *d_gtable_big = nullptr; /* high-cardinality direct kernel table */
  void *d_proj_scratch = nullptr; /* k_decode_multi proj/len/out arrays */
  std::vector<obx_group_row> last_rows; /* full sorted group rows of the
                                           last scan (obx_gpu_agg_fetch) */
  bool lds_ok = false;   /* all blocks 16-B aligned and <= LDS stage size */
  bool in_use = false;
  /* per-(block,col) encoding summary captured at load time for the JIT
     eligibility walk (obx_jit.inc): low nibble = decode class (0 raw,
     1 dict, 2 intdiff, 3 const, 4 slow), bit4 = string; col_cnt = dict
     count capped at 255 */
  std::vector<uint8_t> col_class, col_cnt;
  /* stored skip index: per-(block,col) [min,max] of non-null decoded
     values, captured by k_col_minmax at load (device copy feeds
     k_lower_leaves pruning; host summary feeds the JIT's i64-accumulator
     range eligibility, obx_jit.inc) */
  int64_t *d_minmax = nullptr;
  int64_t col_min[OBX_DEV_MAX_COLS] = {};   /* global over blocks */
  int64_t col_max[OBX_DEV_MAX_COLS] = {};
  bool col_known[OBX_DEV_MAX_COLS] = {};    /* bounds known in EVERY block */
  bool col_dict_every[OBX_DEV_MAX_COLS] = {}; /* class==1 and count<=63
                                                 in every block */
  bool col_dict_any[OBX_DEV_MAX_COLS] = {};   /* dict (incl. char) with
                                                 count<=63 in every block */
  bool col_dict_stable[OBX_DEV_MAX_COLS] = {}; /* dict_any AND the dict
                                                  payload is byte-identical
                                                  in every block (group/ref
                                                  persistence; numeric-only
                                                  uses additionally require
                                                  col_dict_every) */
  bool col_ext_any[OBX_DEV_MAX_COLS] = {};  /* HAS_EXT in any block */
  bool col_raw8_every[OBX_DEV_MAX_COLS] = {}; /* 8-B RAW, 64-bit-aligned,
                                                 no ext, in every block */
  bool col_rangefam_every[OBX_DEV_MAX_COLS] = {}; /* RAW numeric / INTDIFF
                                                     (packed-range lowerable)
                                                     in every block */
  uint32_t col_maxcnt[OBX_DEV_MAX_COLS] = {}; /* max dict count */
  uint32_t col_maxw[OBX_DEV_MAX_COLS] = {};   /* max packed width (bits)
                                                 across blocks; 255 = no
                                                 packed stream (slow enc) */
  uint32_t max_block_rows = 0;
  uint32_t max_block_len = 0;
};

struct obx_gpu_ctx {
  int device = 0;
  hipStream_t stream = nullptr;
  hipEvent_t ev_start = nullptr, ev_stop = nullptr;
  hipEvent_t ev_p0 = nullptr, ev_p1 = nullptr;
  double last_ms = 0.0;
  double last_prep_ms = 0.0;  /* plan upload + per-block filter lowering */
  int last_jit = 0;           /* last scan used the specialized kernel */
  std::vector<obx_handle> handles;
};

extern "C" int obx_gpu_open(int device, obx_gpu_ctx **out) {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess || n <= device) return OBX_NO_GPU;
  auto *ctx = new obx_gpu_ctx();
  ctx->device = device;
  HIP_TRY(hipSetDevice(device));
  HIP_TRY(hipStreamCreate(&ctx->stream));
  HIP_TRY(hipEventCreate(&ctx->ev_start));
  HIP_TRY(hipEventCreate(&ctx->ev_stop));
  HIP_TRY(hipEventCreate(&ctx->ev_p0));
  HIP_TRY(hipEventCreate(&ctx->ev_p1));
  *out = ctx;
  return OBX_SUCCESS;
}

extern "C" int obx_gpu_close(obx_gpu_ctx *ctx) {
  if (!ctx) return OBX_SUCCESS;
  (void)hipSetDevice(ctx->device);
  for (auto &h : ctx->handles) {
    if (!h.in_use) continue;
    (void)hipFree(h.d_buf); (void)hipFree(h.d_blocks); (void)hipFree(h.d_bleaves);
    (void)hipFree(h.d_pleaves); (void)hipFree(h.d_gtable); (void)hipFree(h.d_counters); (void)hipFree(h.d_gtable_big); (void)hipFree(h.d_proj_scratch);
    (void)hipFree(h.d_bitmap); (void)hipFree(h.d_row_ids); (void)hipFree(h.d_blk_counts);
    (void)hipFree(h.d_row_slot); (void)hipFree(h.d_minmax);
    for (auto *p : h.d_decode_out) (void)hipFree(p);
  }
  (void)hipStreamDestroy(ctx->stream);
  (void)hipEventDestroy(ctx->ev_start);
  (void)hipEventDestroy(ctx->ev_stop);
  delete ctx;
  return OBX_SUCCESS;
}

/* Load-time payload checksum (the reference verifies data_checksum_ when a
 * block enters the block cache, before any decode — check_payload_checksum,
 * ob_micro_block_header.cpp:257-271). ob_crc64_sse42 semantics = CRC-32C in
 * a u64 accumulator; the host has the crc32 instruction, same as the
 * reference's hardware path. Independent of the oracle's implementation. */
__attribute__((target("sse4.2")))
static uint64_t host_crc32c(const uint8_t *buf, int64_t len) {
  uint64_t crc = 0;
  int64_t i = 0;
  for (; i + 8 <= len; i += 8) {
    uint64_t w;
    memcpy(&w, buf + i, 8);
    crc = __builtin_ia32_crc32di(crc, w);
  }
  for (; i < len; i++) crc = __builtin_ia32_crc32qi((uint32_t)crc, buf[i]);
  return crc;
}

static uint32_t grid_for(uint32_t n_blocks) {
  uint32_t g = n_blocks < 4096u ? n_blocks : 4096u;
  return g ? g : 1;
}

/* ---- host block parsing (mirrors ObMicroBlockDecoder pointer math,
 * ob_micro_block_decoder.cpp:360-380, and the per-decoder init functions
 * cited in oracle/obx_codec.c) ------------------------------------------- */
static int parse_block(const obx_col_schema *cols, uint16_t n_cols,
                       const uint8_t *block, uint64_t dev_off,
                       uint64_t block_len, uint64_t row_start,
                       dev_block *out) {
  const obx_micro_header *h = (const obx_micro_header *)block;
  if (h->magic != OBX_MICRO_BLOCK_MAGIC) return OBX_INVALID_ARGUMENT;
  if (h->column_count != n_cols) return OBX_INVALID_ARGUMENT;
  if (h->data_zlength > (int64_t)block_len ||
      h->data_zlength < OBX_MICRO_HEADER_SIZE)
    return OBX_INVALID_ARGUMENT;
  if ((int64_t)host_crc32c(block + OBX_MICRO_HEADER_SIZE,
                           (int64_t)h->data_zlength - OBX_MICRO_HEADER_SIZE) !=
      h->data_checksum)
    return OBX_PHYSIC_CHECKSUM_ERROR;
  memset(out, 0, sizeof(*out));
  out->row_start_lo = (uint32_t)row_start;
  out->row_start_hi = (uint32_t)(row_start >> 32);
  out->row_count = h->row_count;
  out->block_len = (uint32_t)block_len;
  out->block_byte = dev_off;
  const uint64_t meta_base =
      dev_off + h->header_size + (uint64_t)n_cols * sizeof(obx_col_header);
  const obx_col_header *chp =
      (const obx_col_header *)(block + h->header_size);
  const uint8_t *meta_host = block + h->header_size +
                             (uint64_t)n_cols * sizeof(obx_col_header);
  const uint8_t evb = obx_hdr_extend_value_bit(h);

  for (uint16_t c = 0; c < n_cols; c++) {
    const obx_col_header *ch = &chp[c];
    dev_col *dc = &out->cols[c];
    const int sc = obx_store_class(ch->obj_type);
    const int64_t tss = obx_type_store_size(ch->obj_type);
    dc->enc = (uint8_t)ch->type;
    dc->datum_len = cols[c].len;
    dc->tss = (uint8_t)(tss > 0 ? tss : cols[c].len);
    dc->ext_width = evb;
    if (sc == OBX_SC_INT) dc->flags |= OBX_DF_SIGNED;
    if (sc == OBX_SC_STRING) dc->flags |= OBX_DF_STRING;
    const uint64_t col_base = meta_base + ch->offset;
    const uint8_t *col_host = meta_host + ch->offset;
    const int has_ext = (ch->attr & OBX_COL_ATTR_HAS_EXTEND_VALUE) ? 1 : 0;
    const int bp = (ch->attr & OBX_COL_ATTR_BIT_PACKING) ? 1 : 0;

    switch (ch->type) {
      case OBX_COL_RAW: {
        if (bp) dc->flags |= OBX_DF_BITPACK;
        dc->width = (uint8_t)ch->length;
        uint64_t ext_bits = has_ext ? (uint64_t)evb * h->row_count : 0;
        if (has_ext) {
          dc->flags |= OBX_DF_HAS_EXT;
          dc->ext_bit = col_base * 8;
        }
        if (bp) {
          dc->data_bit = col_base * 8 + ext_bits;
        } else {
          dc->data_bit = (col_base + (ext_bits + 7) / 8) * 8;
        }
        break;
      }
      case OBX_COL_DICT: {
        const obx_dict_meta *dm = (const obx_dict_meta *)col_host;
        if (bp) dc->flags |= OBX_DF_BITPACK;
        dc->width = dm->row_ref_size;
        dc->count = dm->count;
        dc->entry_len = (uint8_t)dm->data_size;
        dc->dict_byte = col_base + sizeof(obx_dict_meta);
        dc->data_bit = (col_base + ch->length) * 8; /* refs follow meta */
        break;
      }
      case OBX_COL_RLE: {
        const obx_rle_meta *rm = (const obx_rle_meta *)col_host;
        dc->runs = rm->count;
        dc->rib = rm->attr & 7;
        dc->rfb = (rm->attr >> 3) & 7;
        dc->aux_byte = col_base + sizeof(obx_rle_meta);
        const obx_dict_meta *dm =
            (const obx_dict_meta *)(col_host + rm->offset);
        dc->count = dm->count;
        dc->entry_len = (uint8_t)dm->data_size;
        dc->dict_byte = col_base + rm->offset + sizeof(obx_dict_meta);
        break;
      }
      case OBX_COL_CONST: {
        const obx_const_meta *cm = (const obx_const_meta *)col_host;
        dc->runs = cm->count; /* exception count */
        dc->rib = cm->attr & 7;
        dc->rfb = cm->const_ref;
        if (cm->count == 0) {
          if (cm->const_ref > 0) {
            dc->count = 0; /* null-const */
          } else {
            dc->count = 1;
            int64_t cell = (sc == OBX_SC_INT) ? tss : cols[c].len;
            uint64_t v = 0;
            memcpy(&v, col_host + cm->offset, (size_t)cell);
            dc->base = (sc == OBX_SC_INT)
                           ? (int64_t)obx_sign_extend(v, tss, 1)
                           : (sc == OBX_SC_DECIMAL
                                  ? (int64_t)obx_sign_extend(v, cell, 1)
                                  : (int64_t)v);
          }
        } else {
          dc->aux_byte = col_base + sizeof(obx_const_meta);
          const obx_dict_meta *dm =
              (const obx_dict_meta *)(col_host + cm->offset);
          dc->count = dm->count;
          dc->entry_len = (uint8_t)dm->data_size;
          dc->dict_byte = col_base + cm->offset + sizeof(obx_dict_meta);
        }
        break;
      }
      case OBX_COL_INTEGER_BASE_DIFF: {
        const obx_intdiff_meta *im = (const obx_intdiff_meta *)col_host;
        if (bp) dc->flags |= OBX_DF_BITPACK;
        dc->width = im->length;
        uint64_t b = 0;
        memcpy(&b, col_host + sizeof(obx_intdiff_meta), (size_t)tss);
        dc->base = (int64_t)obx_sign_extend(b, tss, sc == OBX_SC_INT);
        uint64_t data0 = col_base + ch->length;
        uint64_t ext_bits = has_ext ? (uint64_t)evb * h->row_count : 0;
        if (has_ext) {
          dc->flags |= OBX_DF_HAS_EXT;
          dc->ext_bit = data0 * 8;
        }
        if (bp) dc->data_bit = data0 * 8 + ext_bits;
        else dc->data_bit = (data0 + (ext_bits + 7) / 8) * 8;
        break;
      }
      case OBX_COL_STRING_PREFIX: {
        const obx_sprefix_meta *pm = (const obx_sprefix_meta *)col_host;
        dc->count = pm->count;
        dc->entry_len = pm->hex_char_cnt;
        dc->rib = pm->pib;
        dc->dict_byte = col_base + sizeof(obx_sprefix_meta); /* hex chars */
        dc->aux_byte = dc->dict_byte + pm->hex_char_cnt;     /* end index */
        /* fixed cell stride: 1 + max suffix (raw or nibble-packed) */
        {
          const uint8_t *ends = col_host + sizeof(obx_sprefix_meta) +
                                pm->hex_char_cnt;
          uint64_t prev = 0, min_pl = ~0ull;
          for (uint32_t j = 0; j < pm->count; j++) {
            uint64_t e2 = 0;
            memcpy(&e2, ends + (size_t)j * pm->pib, pm->pib);
            if (e2 - prev < min_pl) min_pl = e2 - prev;
            prev = e2;
          }
          uint32_t max_suffix = pm->string_size - (uint32_t)min_pl;
          dc->width = (uint8_t)(1 + (pm->hex_char_cnt
                                         ? (max_suffix + 1) / 2
                                         : max_suffix));
        }
        uint64_t data0 = col_base + ch->length;
        uint64_t ext_bits = has_ext ? (uint64_t)evb * h->row_count : 0;
        if (has_ext) {
          dc->flags |= OBX_DF_HAS_EXT;
          dc->ext_bit = data0 * 8;
        }
        dc->data_bit = (data0 + (ext_bits + 7) / 8) * 8;
        break;
      }
      case OBX_COL_SUBSTR: {
        const obx_substr_meta *sm3 = (const obx_substr_meta *)col_host;
        if (sm3->ref_col >= n_cols) return OBX_INVALID_ARGUMENT;
        dc->runs = sm3->exc_cnt;
        dc->rib = sm3->rib;
        dc->width = (uint8_t)sm3->ref_col; /* ref column index */
        dc->base = sm3->start_pos;
        dc->dict_byte = col_base + sizeof(obx_substr_meta); /* exc rids */
        dc->aux_byte = dc->dict_byte + (uint64_t)sm3->exc_cnt * sm3->rib;
        break;
      }
      case OBX_COL_EQUAL: {
        const obx_coleq_meta *em = (const obx_coleq_meta *)col_host;
        if (em->ref_col >= n_cols) return OBX_INVALID_ARGUMENT;
        dc->runs = em->exc_cnt;
        dc->rib = em->rib;
        dc->width = (uint8_t)em->ref_col; /* ref column index */
        dc->dict_byte = col_base + sizeof(obx_coleq_meta); /* exc row_ids */
        dc->aux_byte = dc->dict_byte + (uint64_t)em->exc_cnt * em->rib;
        break;
      }
      case OBX_COL_HEX_PACKING: {
        const obx_hex_meta *hm = (const obx_hex_meta *)col_host;
        dc->width = (uint8_t)((hm->string_size + 1) / 2); /* row stride B */
        dc->count = hm->char_cnt;
        dc->dict_byte = col_base + sizeof(obx_hex_meta); /* char array */
        uint64_t data0 = col_base + ch->length;
        uint64_t ext_bits = has_ext ? (uint64_t)evb * h->row_count : 0;
        if (has_ext) {
          dc->flags |= OBX_DF_HAS_EXT;
          dc->ext_bit = data0 * 8;
        }
        dc->data_bit = (data0 + (ext_bits + 7) / 8) * 8;
        break;
      }
      case OBX_COL_STRING_DIFF: {
        const obx_sdiff_meta *sm = (const obx_sdiff_meta *)col_host;
        dc->runs = sm->diff_desc_cnt;
        dc->entry_len = sm->hex_char_cnt;
        dc->dict_byte = col_base + sizeof(obx_sdiff_meta); /* descs */
        const uint8_t *descs = col_host + sizeof(obx_sdiff_meta);
        uint32_t diff_len = 0;
        for (uint32_t i = 0; i < sm->diff_desc_cnt; i++)
          if (descs[i] & 1) diff_len += descs[i] >> 1;
        dc->rib = (uint8_t)diff_len;
        dc->width = (uint8_t)(sm->hex_char_cnt ? (diff_len + 1) / 2
                                               : diff_len);
        dc->aux_byte = dc->dict_byte + sm->diff_desc_cnt +
                       sm->hex_char_cnt; /* common bytes */
        uint64_t data0 = col_base + ch->length;
        uint64_t ext_bits = has_ext ? (uint64_t)evb * h->row_count : 0;
        if (has_ext) {
          dc->flags |= OBX_DF_HAS_EXT;
          dc->ext_bit = data0 * 8;
        }
        dc->data_bit = (data0 + (ext_bits + 7) / 8) * 8;
        break;
      }
      default:
        return OBX_NOT_SUPPORTED;
    }
  }
  return OBX_SUCCESS;
}

extern "C" int obx_gpu_load_blocks(obx_gpu_ctx *ctx, const obx_blockset *bs) {
  if (!ctx || !bs) return OBX_INVALID_ARGUMENT;
  if (bs->n_cols > OBX_DEV_MAX_COLS) return OBX_NOT_SUPPORTED;
  for (uint16_t c = 0; c < bs->n_cols; c++) {
    /* decimal scale bounds the P10 tables (reference: decimal-int
       precision <= 18 digits on this path) */
    if (bs->cols[c].scale < 0 || bs->cols[c].scale > 18)
      return OBX_NOT_SUPPORTED;
  }
  HIP_TRY(hipSetDevice(ctx->device));
  obx_handle h;
  h.n_blocks = bs->n_blocks;
  h.n_cols = bs->n_cols;
  h.total_bytes = bs->block_offsets[bs->n_blocks];
  memcpy(h.cols, bs->cols, sizeof(obx_col_schema) * bs->n_cols);

  std::vector<dev_block> blocks(bs->n_blocks);
  uint64_t row_start = 0;
  bool lds_ok = true;
  for (uint32_t b = 0; b < bs->n_blocks; b++) {
    uint64_t blen = bs->block_offsets[b + 1] - bs->block_offsets[b];
    int rc = parse_block(bs->cols, bs->n_cols, bs->data + bs->block_offsets[b],
                         bs->block_offsets[b], blen, row_start, &blocks[b]);
    if (rc != OBX_SUCCESS) return rc;
    if ((bs->block_offsets[b] & 15) || blen + 24 > OBX_LDS_STAGE_BYTES ||
        blocks[b].row_count > 4096 /* OBX_MAX_BLOCK_ROWS */)
      lds_ok = false;
    for (uint16_t c = 0; c < bs->n_cols; c++) {
      const dev_col &dc = blocks[b].cols[c];
      uint8_t cls;
      switch (dc.enc) {
        case OBX_D_RAW: cls = 0; break;
        case OBX_D_DICT: cls = 1; break;
        case OBX_D_INTDIFF: cls = 2; break;
        case OBX_D_CONST: cls = dc.runs == 0 ? 3 : 4; break;
        case OBX_D_EQUAL: case OBX_D_SUBSTR:
          cls = 5; break; /* span: no device group keys */
        default: cls = 4; break;
      }
      if (dc.flags & OBX_DF_STRING) cls |= 0x10;
      h.col_class.push_back(cls);
      h.col_cnt.push_back((uint8_t)(dc.count > 254 ? 255 : dc.count));
      if (b == 0) {
        h.col_dict_every[c] = true;
        h.col_dict_any[c] = true;
        h.col_raw8_every[c] = true;
        h.col_rangefam_every[c] = true;
        h.col_maxcnt[c] = 0;
      }
      if ((cls & 0xF) != 1 || (cls & 0x10) || dc.count > 63)
        h.col_dict_every[c] = false;
      if ((cls & 0xF) != 1 || dc.count > 63)
        h.col_dict_any[c] = false; /* dict (possibly char) everywhere */
      if (!(dc.enc == OBX_D_RAW && !(dc.flags & OBX_DF_BITPACK) &&
            !(dc.flags & OBX_DF_STRING) && !(dc.flags & OBX_DF_HAS_EXT) &&
            dc.width == 8 && (dc.data_bit & 63) == 0))
        h.col_raw8_every[c] = false;
      if (!((dc.enc == OBX_D_RAW || dc.enc == OBX_D_INTDIFF) &&
            !(dc.flags & OBX_DF_STRING) && !(dc.flags & OBX_DF_HAS_EXT)))
        h.col_rangefam_every[c] = false;
      if (dc.count > h.col_maxcnt[c]) h.col_maxcnt[c] = dc.count;
      if (dc.flags & OBX_DF_HAS_EXT) h.col_ext_any[c] = true;
      {
        uint32_t w = 255; /* no packed stream: blocks multi-row reads */
        if (dc.enc == OBX_D_RAW || dc.enc == OBX_D_DICT ||
            dc.enc == OBX_D_INTDIFF)
          w = (dc.flags & OBX_DF_BITPACK) ? dc.width : (uint32_t)dc.width * 8;
        if (w > h.col_maxw[c]) h.col_maxw[c] = w;
      }
    }
    if (blocks[b].row_count > h.max_block_rows)
      h.max_block_rows = blocks[b].row_count;
    if (blen > h.max_block_len) h.max_block_len = (uint32_t)blen;
    row_start += blocks[b].row_count;
  }
  h.total_rows = row_start;
  h.lds_ok = lds_ok;
  /* dict stability: identical dict payload in every block enables the
     JIT's persistent (whole-kernel) histograms/value tables */
  for (uint16_t c = 0; c < bs->n_cols; c++) {
    h.col_dict_stable[c] = h.col_dict_any[c];
    if (!h.col_dict_stable[c]) continue;
    const dev_col &d0 = blocks[0].cols[c];
    size_t dlen = (size_t)d0.count * d0.entry_len;
    for (uint32_t b = 1; b < bs->n_blocks && h.col_dict_stable[c]; b++) {
      const dev_col &db = blocks[b].cols[c];
      if (db.count != d0.count || db.entry_len != d0.entry_len ||
          memcmp(bs->data + db.dict_byte, bs->data + d0.dict_byte, dlen))
        h.col_dict_stable[c] = false;
    }
  }

  HIP_TRY(hipMalloc(&h.d_buf, h.total_bytes + 64));
  HIP_TRY(hipMemset(h.d_buf + h.total_bytes, 0, 64));
  HIP_TRY(hipMemcpy(h.d_buf, bs->data, h.total_bytes, hipMemcpyHostToDevice));
  HIP_TRY(hipMalloc(&h.d_blocks, sizeof(dev_block) * bs->n_blocks));
  HIP_TRY(hipMemcpy(h.d_blocks, blocks.data(),
                    sizeof(dev_block) * bs->n_blocks, hipMemcpyHostToDevice));
  HIP_TRY(hipMalloc(&h.d_counters, 16 * sizeof(unsigned long long)));
  HIP_TRY(hipMalloc(&h.d_gtable, sizeof(gslot) * OBX_GTABLE_SLOTS));
  HIP_TRY(hipMalloc(&h.d_pleaves, sizeof(dev_leaf) * OBX_DEV_MAX_LEAVES));

  /* stored skip index: capture per-(block,col) min/max once at load
     (k_col_minmax; outside every timed region) and summarize per column
     for the JIT's range eligibility */
  {
    uint64_t nmm = (uint64_t)bs->n_blocks * bs->n_cols * 2;
    HIP_TRY(hipMalloc(&h.d_minmax, nmm * sizeof(int64_t)));
    hipLaunchKernelGGL(k_col_minmax, dim3(grid_for(bs->n_blocks)),
                       dim3(OBX_WG_HOST), 0, nullptr, h.d_buf, h.d_blocks,
                       bs->n_blocks, (uint32_t)bs->n_cols, h.d_minmax);
    std::vector<int64_t> mm(nmm);
    HIP_TRY(hipMemcpy(mm.data(), h.d_minmax, nmm * sizeof(int64_t),
                      hipMemcpyDeviceToHost));
    for (uint16_t c = 0; c < bs->n_cols; c++) {
      int64_t gmin = INT64_MAX, gmax = INT64_MIN;
      bool known = true;
      for (uint32_t b = 0; b < bs->n_blocks; b++) {
        int64_t mn = mm[2 * ((uint64_t)b * bs->n_cols + c)];
        int64_t mx = mm[2 * ((uint64_t)b * bs->n_cols + c) + 1];
        if (mn == INT64_MIN && mx == INT64_MAX) { known = false; break; }
        if (mn <= mx) { /* empty (all-null) blocks don't widen */
          if (mn < gmin) gmin = mn;
          if (mx > gmax) gmax = mx;
        }
      }
      h.col_known[c] = known;
      h.col_min[c] = gmin;
      h.col_max[c] = gmax;
    }
  }
  h.in_use = true;
  ctx->handles.push_back(h);
  return (int)ctx->handles.size() - 1;
}

extern "C" int obx_gpu_free_blocks(obx_gpu_ctx *ctx, int handle) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size())
    return OBX_INVALID_ARGUMENT;
  HIP_TRY(hipSetDevice(ctx->device));
  obx_handle &h = ctx->handles[handle];
  if (!h.in_use) return OBX_INVALID_ARGUMENT;
  (void)hipFree(h.d_buf); (void)hipFree(h.d_blocks); (void)hipFree(h.d_bleaves);
  (void)hipFree(h.d_pleaves); (void)hipFree(h.d_gtable); (void)hipFree(h.d_counters); (void)hipFree(h.d_gtable_big); (void)hipFree(h.d_proj_scratch);
  (void)hipFree(h.d_bitmap); (void)hipFree(h.d_row_ids); (void)hipFree(h.d_blk_counts);
  (void)hipFree(h.d_row_slot); (void)hipFree(h.d_minmax);
  for (auto *&p : h.d_decode_out) { (void)hipFree(p); p = nullptr; }
  h = obx_handle();
  return OBX_SUCCESS;
}

/* build plan header + device leaves (pure: no HIP calls) */
static int build_plan(const obx_handle &h, const obx_filter_desc *filter,
                      const obx_agg_desc *agg, dev_plan_hdr &ph,
                      dev_leaf pl[OBX_DEV_MAX_LEAVES]) {
  memset(&ph, 0, sizeof(ph));
  memset(pl, 0, sizeof(dev_leaf) * OBX_DEV_MAX_LEAVES);
  uint16_t nl = filter ? filter->n_leaves : 0;
  if (nl > OBX_DEV_MAX_LEAVES) return OBX_INVALID_ARGUMENT;
  ph.n_leaves = nl;
  if (filter && filter->n_prog) {
    /* validate the combine program (operand range, stack discipline) */
    int sp = 0;
    for (int p = 0; p < filter->n_prog; p++) {
      uint8_t t = filter->prog[p];
      if (t < nl) sp++;
      else if ((t == OBX_TOK_AND || t == OBX_TOK_OR) && sp >= 2) sp--;
      else return OBX_INVALID_ARGUMENT;
      if (sp > 8) return OBX_INVALID_ARGUMENT;
    }
    if (sp != 1) return OBX_INVALID_ARGUMENT;
    ph.n_prog = filter->n_prog;
    memcpy(ph.prog, filter->prog, filter->n_prog);
  }
  for (uint16_t i = 0; i < nl; i++) {
    const obx_filter_leaf *lf = &filter->leaves[i];
    if (lf->col >= h.n_cols) return OBX_INVALID_ARGUMENT;
    pl[i].col = lf->col;
    pl[i].op = lf->op;
    pl[i].vlo = lf->lo;
    pl[i].vhi = lf->hi;
    pl[i].n_in = lf->n_in;
    for (int j = 0; j < lf->n_in && j < 8; j++) pl[i].in_list[j] = lf->in_list[j];
    if (lf->op == OBX_OP_BLACK) {
      /* stack-discipline validation mirroring the oracle's
         obx__bprog_valid */
      if (lf->n_bprog == 0 || lf->n_bprog > OBX_BX_MAX_PROG ||
          lf->n_bcols == 0 || lf->n_bcols > OBX_BX_MAX_COLS)
        return OBX_INVALID_ARGUMENT;
      int sp = 0;
      for (int p = 0; p < lf->n_bprog; p++) {
        uint8_t op2 = lf->bprog[p];
        if (op2 < 0x40) {
          if (op2 >= lf->n_bcols) return OBX_INVALID_ARGUMENT;
          sp++;
        } else if (op2 < 0x50) {
          if ((op2 & 0x0F) >= OBX_BX_MAX_CONST) return OBX_INVALID_ARGUMENT;
          sp++;
        } else if (op2 == 0x54 || op2 == 0x72) {
          if (sp < 1) return OBX_INVALID_ARGUMENT;
        } else if ((op2 >= 0x50 && op2 <= 0x53) || op2 == 0x55 ||
                   (op2 >= 0x60 && op2 <= 0x65) || op2 == 0x70 ||
                   op2 == 0x71) {
          if (sp < 2) return OBX_INVALID_ARGUMENT;
          sp--;
        } else {
          return OBX_INVALID_ARGUMENT;
        }
        if (sp > 8) return OBX_INVALID_ARGUMENT;
      }
      if (sp != 1) return OBX_INVALID_ARGUMENT;
      for (int j = 0; j < lf->n_bcols; j++) {
        if (lf->bcols[j] >= h.n_cols) return OBX_INVALID_ARGUMENT;
        pl[i].bcols[j] = lf->bcols[j];
      }
      pl[i].n_bcols = lf->n_bcols;
      pl[i].n_bprog = lf->n_bprog;
      memcpy(pl[i].bprog, lf->bprog, lf->n_bprog);
      memcpy(pl[i].bconst, lf->bconst, sizeof(pl[i].bconst));
      pl[i].col = lf->bcols[0]; /* pruning/lowering anchor */
      continue;
    }
    /* char columns compare byte-lexicographically: order-map the operands
       exactly like the oracle's char_key (low len LE bytes -> BE int) */
    if (obx_store_class(h.cols[lf->col].obj_type) == OBX_SC_STRING) {
      uint32_t len = h.cols[lf->col].len;
      auto ck = [len](int64_t x) -> int64_t {
        return (int64_t)(__builtin_bswap64((uint64_t)x) >> (8 * (8 - len)));
      };
      pl[i].char_len = (uint8_t)len;
      pl[i].vlo = ck(lf->lo);
      pl[i].vhi = ck(lf->hi);
      for (int j = 0; j < lf->n_in && j < 8; j++)
        pl[i].in_list[j] = ck(lf->in_list[j]);
    }
  }
  /* AND-only plans: merge a lower-bound and an upper-bound leaf on the
     same column into one OP_BT leaf (the reference's range-node build,
     ObWhiteFilterExecutor; one pass instead of two in the kernels) */
  if (filter && !filter->n_prog && nl > 1) {
    for (uint16_t i = 0; i < nl; i++) {
      if (pl[i].op != OBX_OP_GE && pl[i].op != OBX_OP_GT) continue;
      for (uint16_t k = 0; k < nl; k++) {
        if (k == i || pl[k].col != pl[i].col) continue;
        if (pl[k].op != OBX_OP_LE && pl[k].op != OBX_OP_LT) continue;
        int64_t lo = pl[i].vlo, hi = pl[k].vlo;
        if (pl[i].op == OBX_OP_GT) {
          if (lo == INT64_MAX) continue;
          lo++;
        }
        if (pl[k].op == OBX_OP_LT) {
          if (hi == INT64_MIN) continue;
          hi--;
        }
        pl[i].op = OBX_OP_BT;
        pl[i].vlo = lo;
        pl[i].vhi = hi;
        /* drop leaf k */
        for (uint16_t m = k; m + 1 < nl; m++) pl[m] = pl[m + 1];
        nl--;
        ph.n_leaves = nl;
        if (k < i) i--;
        i--; /* re-examine the merged leaf (another pair may exist) */
        break;
      }
    }
  }

  /* resolve needed value slots */
  int slot_of_col[64];
  for (int i = 0; i < 64; i++) slot_of_col[i] = -1;
  auto need = [&](uint16_t c) -> uint8_t {
    if (slot_of_col[c] < 0) {
      if (ph.n_need >= OBX_DEV_MAX_NEED) return 0xFF;
      slot_of_col[c] = ph.n_need;
      ph.need_cols[ph.n_need++] = c;
    }
    return (uint8_t)slot_of_col[c];
  };
  if (agg) {
    if (agg->n_group_cols > 2 || agg->n_aggs > OBX_DEV_MAX_AGGS)
      return OBX_INVALID_ARGUMENT;
    ph.n_group_cols = agg->n_group_cols;
    for (int g = 0; g < agg->n_group_cols; g++) {
      if (agg->group_cols[g] >= h.n_cols) return OBX_INVALID_ARGUMENT;
      ph.group_idx[g] = need(agg->group_cols[g]);
      ph.group_len[g] = h.cols[agg->group_cols[g]].len;
    }
    ph.n_aggs = agg->n_aggs;
    for (int a = 0; a < agg->n_aggs; a++) {
      const obx_agg_expr *e = &agg->aggs[a];
      dev_agg *da = &ph.aggs[a];
      da->kind = e->kind;
      da->ia = da->ib = da->ic = 0xFF;
      if (e->col_a != UINT16_MAX && e->col_a >= h.n_cols)
        return OBX_INVALID_ARGUMENT;
      if (e->col_a != UINT16_MAX) da->ia = need(e->col_a);
      if (e->kind == OBX_AGG_SUM_PROD2 || e->kind == OBX_AGG_SUM_PROD3 ||
          e->kind == OBX_AGG_SUM_MUL) {
        if (e->col_b >= h.n_cols) return OBX_INVALID_ARGUMENT;
        da->ib = need(e->col_b);
        static const int64_t P10[19] = {1ll,10ll,100ll,1000ll,10000ll,
          100000ll,1000000ll,10000000ll,100000000ll,1000000000ll,
          10000000000ll,100000000000ll,1000000000000ll,10000000000000ll,
          100000000000000ll,1000000000000000ll,10000000000000000ll,
          100000000000000000ll,1000000000000000000ll};
        da->one_b = P10[h.cols[e->col_b].scale];
        if (e->kind == OBX_AGG_SUM_PROD3) {
          if (e->col_c >= h.n_cols) return OBX_INVALID_ARGUMENT;
          da->ic = need(e->col_c);
          da->one_c = P10[h.cols[e->col_c].scale];
        }
      }
    }
    /* group keys decode via col_value (no block context): COLUMN_EQUAL
       group columns would misdecode — reject them (value/filter columns
       on COLUMN_EQUAL are fully supported via col_value2) */
    if (!h.col_class.empty()) {
      for (uint32_t g = 0; g < ph.n_group_cols; g++) {
        uint16_t col = ph.need_cols[ph.group_idx[g]];
        for (uint32_t b2 = 0; b2 < h.n_blocks; b2++)
          if ((h.col_class[(size_t)b2 * h.n_cols + col] & 0xF) == 5)
            return OBX_NOT_SUPPORTED;
      }
    }
  }
  /* group aggregates into row passes: aggs whose inputs fit in <=3 shared
     decoded columns run together (COUNT(*) needs no pass — filled from the
     group counts at flush) */
  if (agg) {
    bool used[OBX_DEV_MAX_AGGS] = {};
    ph.n_passes = 0;
    for (int a = 0; a < agg->n_aggs; a++) {
      if (used[a]) continue;
      const dev_agg *da = &ph.aggs[a];
      if (da->kind == OBX_AGG_COUNT && da->ia == 0xFF) { used[a] = true; continue; }
      dev_pass *pp = &ph.passes[ph.n_passes++];
      memset(pp, 0, sizeof(*pp));
      auto col_sel = [&](uint8_t need_idx) -> int {
        if (need_idx == 0xFF) return -1;
        for (int k = 0; k < pp->n_cols; k++)
          if (pp->cols[k] == need_idx) return k;
        if (pp->n_cols >= 3) return -2; /* no room */
        pp->cols[pp->n_cols] = need_idx;
        return pp->n_cols++;
      };
      auto try_add = [&](int idx) -> bool {
        const dev_agg *d = &ph.aggs[idx];
        uint8_t save_n = pp->n_cols;
        int sa = col_sel(d->ia), sb = col_sel(d->ib), sc = col_sel(d->ic);
        if (sa == -2 || sb == -2 || sc == -2) { pp->n_cols = save_n; return false; }
        dev_pass_agg *pa = &pp->aggs[pp->n_aggs++];
        pa->kind = d->kind;
        pa->agg_idx = (uint8_t)idx;
        pa->sa = (uint8_t)(sa < 0 ? 0xFF : sa);
        pa->sb = (uint8_t)(sb < 0 ? 0xFF : sb);
        pa->sc = (uint8_t)(sc < 0 ? 0xFF : sc);
        pa->one_b = d->one_b;
        pa->one_c = d->one_c;
        return true;
      };
      try_add(a);
      used[a] = true;
      for (int b2 = a + 1; b2 < agg->n_aggs; b2++) {
        if (used[b2]) continue;
        const dev_agg *d2 = &ph.aggs[b2];
        if (d2->kind == OBX_AGG_COUNT && d2->ia == 0xFF) { used[b2] = true; continue; }
        if (try_add(b2)) used[b2] = true;
      }
    }
  }

  return OBX_SUCCESS;
}

/* build the plan + upload leaves + lower per-block tests on device */
static int prep_query(obx_gpu_ctx *ctx, obx_handle &h,
                      const obx_filter_desc *filter, const obx_agg_desc *agg,
                      dev_plan_hdr &ph, dev_leaf *pl_out = nullptr) {
  dev_leaf pl[OBX_DEV_MAX_LEAVES];
  int rc = build_plan(h, filter, agg, ph, pl);
  if (pl_out) memcpy(pl_out, pl, sizeof(pl));
  if (rc != OBX_SUCCESS) return rc;
  uint16_t nl = ph.n_leaves;
  /* upload plan leaves + lower per-block tests on device */
  HIP_TRY(hipEventRecord(ctx->ev_p0, ctx->stream));
  HIP_TRY(hipMemcpyAsync(h.d_pleaves, pl, sizeof(pl), hipMemcpyHostToDevice,
                         ctx->stream));
  if (nl > 0) {
    uint64_t needed = (uint64_t)h.n_blocks * nl;
    if (h.bleaves_cap < needed) {
      (void)hipFree(h.d_bleaves);
      HIP_TRY(hipMalloc(&h.d_bleaves, needed * sizeof(blk_leaf)));
      h.bleaves_cap = needed;
    }
    uint32_t total = (uint32_t)needed;
    uint32_t grid = (total + OBX_WG_HOST - 1) / OBX_WG_HOST;
    hipLaunchKernelGGL(k_lower_leaves, dim3(grid), dim3(OBX_WG_HOST), 0, ctx->stream,
                       h.d_buf, h.d_blocks, h.n_blocks, h.d_pleaves, nl,
                       (uint32_t)h.n_cols, (const int64_t *)h.d_minmax,
                       h.d_bleaves);
  }
  HIP_TRY(hipEventRecord(ctx->ev_p1, ctx->stream));
  return OBX_SUCCESS;
}

#include "obx_jit.inc"


extern "C" int obx_gpu_filter(obx_gpu_ctx *ctx, int handle,
                              const obx_filter_desc *filter,
                              int want_row_ids) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size())
    return OBX_INVALID_ARGUMENT;
  HIP_TRY(hipSetDevice(ctx->device));
  obx_handle &h = ctx->handles[handle];
  dev_plan_hdr ph;
  dev_leaf plv[OBX_DEV_MAX_LEAVES];
  int rc = prep_query(ctx, h, filter, nullptr, ph, plv);
  if (rc != OBX_SUCCESS) return rc;
  ctx->last_jit = 0;
  uint64_t bm_words = (h.total_rows + 63) / 64 + 1;
  if (!h.d_bitmap) HIP_TRY(hipMalloc(&h.d_bitmap, bm_words * 8));
  HIP_TRY(hipMemsetAsync(h.d_bitmap, 0, bm_words * 8, ctx->stream));
  const bool no_bitmap = getenv("OBX_NO_BITMAP") != nullptr; /* perf A/B */
  if (want_row_ids) {
    if (!h.d_row_ids)
      HIP_TRY(hipMalloc(&h.d_row_ids, h.total_rows * sizeof(int32_t)));
    if (!h.d_blk_counts)
      HIP_TRY(hipMalloc(&h.d_blk_counts, h.n_blocks * sizeof(uint32_t)));
    /* skipped blocks (NONE verdict) write nothing: stale counts from a
       previous filter would survive (caught by the zero-survivor edge
       test) */
    HIP_TRY(hipMemsetAsync(h.d_blk_counts, 0,
                           h.n_blocks * sizeof(uint32_t), ctx->stream));
  }
  HIP_TRY(hipMemsetAsync(h.d_counters, 0, 16 * 8, ctx->stream));
  /* bitmap-only filters take the specialized wave-per-block kernel
     (direct global reads of just the leaf streams; obx_jit_v2.inc) */
  jit_entry *fje = jit_prepare_filter(h, ph, plv, want_row_ids);
  ctx->last_jit = fje ? 2 : 0;
  HIP_TRY(hipEventRecord(ctx->ev_start, ctx->stream));
  if (fje) {
    /* ~1-2k WGs measured optimal: larger grids thrash concurrent DMA
       streams, and the striped-counter tail scales with grid anyway */
    uint32_t fgrid = h.n_blocks / 8;
    if (fgrid < 512) fgrid = h.n_blocks < 512 ? (h.n_blocks ? h.n_blocks : 1)
                                              : 512;
    if (fgrid > 2048) fgrid = 2048;
    if (const char *fg = getenv("OBX_JIT_FGRID")) {
      int g = atoi(fg);
      if (g > 0 && g <= 65535) fgrid = (uint32_t)g;
    }
    void *rid = want_row_ids ? (void *)h.d_row_ids : nullptr;
    void *bcn = want_row_ids ? (void *)h.d_blk_counts : nullptr;
    void *args[9] = {&h.d_buf, &h.d_blocks, &h.n_blocks, &h.d_pleaves,
                     &h.d_bleaves, &h.d_bitmap, &rid, &bcn, &h.d_counters};
    if (hipModuleLaunchKernel(fje->fn, fgrid, 1, 1,
                              OBX_WG_HOST, 1, 1, 0, ctx->stream, args,
                              nullptr) != hipSuccess)
      return OBX_INTERNAL_ERROR;
  } else {
    auto kfn = h.lds_ok ? (ph.n_prog ? k_filter_prog_lds : k_filter_lds)
                        : (ph.n_prog ? k_filter_prog : k_filter);
    hipLaunchKernelGGL(kfn, dim3(grid_for(h.n_blocks)), dim3(OBX_WG_HOST), 0,
                       ctx->stream, h.d_buf, h.d_blocks, h.n_blocks,
                       h.d_pleaves, h.d_bleaves, ph,
                       (h.lds_ok && no_bitmap) ? nullptr : h.d_bitmap,
                       want_row_ids ? h.d_row_ids : nullptr,
                       want_row_ids ? h.d_blk_counts : nullptr, h.d_counters);
  }
  HIP_TRY(hipEventRecord(ctx->ev_stop, ctx->stream));
  HIP_TRY(hipStreamSynchronize(ctx->stream));
  float ms = 0;
  HIP_TRY(hipEventElapsedTime(&ms, ctx->ev_start, ctx->ev_stop));
  ctx->last_ms = ms;
  HIP_TRY(hipEventElapsedTime(&ms, ctx->ev_p0, ctx->ev_p1));
  ctx->last_prep_ms = ms;
  unsigned long long cnt[16];
  HIP_TRY(hipMemcpy(cnt, h.d_counters, 16 * 8, hipMemcpyDeviceToHost));
  /* slot 0: legacy kernels; slots 8..15: JIT filter's striped WG sums */
  h.last_survivors = cnt[0];
  for (int i2 = 8; i2 < 16; i2++) h.last_survivors += cnt[i2];
  return OBX_SUCCESS;
}

extern "C" int obx_gpu_fetch_bitmap(obx_gpu_ctx *ctx, int handle, uint8_t *out,
                                    int64_t cap) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size())
    return OBX_INVALID_ARGUMENT;
  obx_handle &h = ctx->handles[handle];
  if (!h.d_bitmap) return OBX_INVALID_ARGUMENT;
  int64_t bytes = (int64_t)((h.total_rows + 7) / 8);
  if (cap < bytes) return OBX_BUF_NOT_ENOUGH;
  HIP_TRY(hipMemcpy(out, h.d_bitmap, bytes, hipMemcpyDeviceToHost));
  return OBX_SUCCESS;
}

extern "C" int obx_gpu_fetch_row_ids(obx_gpu_ctx *ctx, int handle,
                                     int32_t *out, int64_t cap,
                                     uint64_t *n_out) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size())
    return OBX_INVALID_ARGUMENT;
  obx_handle &h = ctx->handles[handle];
  if (!h.d_row_ids) return OBX_INVALID_ARGUMENT;
  if (n_out) *n_out = h.last_survivors;
  if (out) {
    if (cap < (int64_t)h.total_rows) return OBX_BUF_NOT_ENOUGH;
    HIP_TRY(hipMemcpy(out, h.d_row_ids, h.total_rows * sizeof(int32_t),
                      hipMemcpyDeviceToHost));
  }
  return OBX_SUCCESS;
}

extern "C" int obx_gpu_fetch_blk_counts(obx_gpu_ctx *ctx, int handle,
                                        uint32_t *out, int64_t cap) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size())
    return OBX_INVALID_ARGUMENT;
  obx_handle &h = ctx->handles[handle];
  if (!h.d_blk_counts) return OBX_INVALID_ARGUMENT;
  if (cap < (int64_t)h.n_blocks) return OBX_BUF_NOT_ENOUGH;
  HIP_TRY(hipMemcpy(out, h.d_blk_counts, h.n_blocks * 4,
                    hipMemcpyDeviceToHost));
  return OBX_SUCCESS;
}

extern "C" int obx_gpu_decode(obx_gpu_ctx *ctx, int handle,
                              const uint16_t *proj_cols, uint16_t n_proj) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size())
    return OBX_INVALID_ARGUMENT;
  HIP_TRY(hipSetDevice(ctx->device));
  obx_handle &h = ctx->handles[handle];
  HIP_TRY(hipEventRecord(ctx->ev_start, ctx->stream));
  if (h.lds_ok && n_proj > 1 && n_proj <= OBX_DEV_MAX_COLS) {
    /* one block stage serves every projected column */
    uint16_t pc[OBX_DEV_MAX_COLS];
    uint8_t ln[OBX_DEV_MAX_COLS];
    uint8_t *po[OBX_DEV_MAX_COLS];
    for (uint16_t i = 0; i < n_proj; i++) {
      uint16_t c = proj_cols[i];
      if (c >= h.n_cols) return OBX_INVALID_ARGUMENT;
      if (!h.d_decode_out[c])
        HIP_TRY(hipMalloc(&h.d_decode_out[c], h.total_rows * h.cols[c].len));
      pc[i] = c;
      ln[i] = (uint8_t)h.cols[c].len;
      po[i] = h.d_decode_out[c];
    }
    if (!h.d_proj_scratch)
      HIP_TRY(hipMalloc(&h.d_proj_scratch,
                        OBX_DEV_MAX_COLS * (8 + 2 + 1) + 16));
    /* pointers first (8-aligned base), then u16 cols, then u8 lens */
    uint8_t *base = (uint8_t *)h.d_proj_scratch;
    uint8_t *outp = base;
    uint8_t *pcp = base + OBX_DEV_MAX_COLS * 8;
    uint8_t *lnp = pcp + OBX_DEV_MAX_COLS * 2;
    /* synchronous copies: the sources are stack arrays */
    HIP_TRY(hipMemcpy(outp, po, sizeof(void *) * n_proj,
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(pcp, pc, sizeof(uint16_t) * n_proj,
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(lnp, ln, n_proj, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_decode_multi, dim3(grid_for(h.n_blocks)),
                       dim3(OBX_WG_HOST), 0, ctx->stream, h.d_buf,
                       h.d_blocks, h.n_blocks, (const uint16_t *)pcp,
                       lnp, (uint32_t)n_proj, (uint8_t *const *)outp,
                       (uint8_t *const *)nullptr);
  } else {
  for (uint16_t i = 0; i < n_proj; i++) {
    uint16_t c = proj_cols[i];
    if (c >= h.n_cols) return OBX_INVALID_ARGUMENT;
    if (!h.d_decode_out[c])
      HIP_TRY(hipMalloc(&h.d_decode_out[c], h.total_rows * h.cols[c].len));
    auto kfn = h.lds_ok ? k_decode_lds : k_decode;
    hipLaunchKernelGGL(kfn, dim3(grid_for(h.n_blocks)), dim3(OBX_WG_HOST), 0,
                       ctx->stream, h.d_buf, h.d_blocks, h.n_blocks,
                       (uint32_t)c, (uint32_t)h.cols[c].len,
                       h.d_decode_out[c], (uint8_t *)nullptr);
  }
  }
  HIP_TRY(hipEventRecord(ctx->ev_stop, ctx->stream));
  HIP_TRY(hipStreamSynchronize(ctx->stream));
  float ms = 0;
  HIP_TRY(hipEventElapsedTime(&ms, ctx->ev_start, ctx->ev_stop));
  ctx->last_ms = ms; /* per-operator monitoring, as the scan verbs */
  return OBX_SUCCESS;
}

extern "C" int obx_gpu_fetch_col(obx_gpu_ctx *ctx, int handle, uint16_t col,
                                 uint8_t *out, int64_t cap) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size())
    return OBX_INVALID_ARGUMENT;
  obx_handle &h = ctx->handles[handle];
  if (col >= h.n_cols || !h.d_decode_out[col]) return OBX_INVALID_ARGUMENT;
  int64_t bytes = (int64_t)h.total_rows * h.cols[col].len;
  if (cap < bytes) return OBX_BUF_NOT_ENOUGH;
  HIP_TRY(hipMemcpy(out, h.d_decode_out[col], bytes, hipMemcpyDeviceToHost));
  return OBX_SUCCESS;
}

extern "C" int obx_gpu_scan_filter_agg(obx_gpu_ctx *ctx, int handle,
                                       const obx_filter_desc *filter,
                                       const obx_agg_desc *agg,
                                       obx_agg_result *out) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size() || !out)
    return OBX_INVALID_ARGUMENT;
  HIP_TRY(hipSetDevice(ctx->device));
  obx_handle &h = ctx->handles[handle];
  h.last_rows.clear(); /* no stale pagination after a failed scan */
  dev_plan_hdr ph;
  dev_leaf plv[OBX_DEV_MAX_LEAVES];
  int rc = prep_query(ctx, h, filter, agg, ph, plv);
  if (rc != OBX_SUCCESS) return rc;
  ctx->last_jit = 0;

  /* init global table (min/max cells need INT64_MAX/MIN) */
  std::vector<gslot> init(OBX_GTABLE_SLOTS);
  memset(init.data(), 0, sizeof(gslot) * OBX_GTABLE_SLOTS);
  for (auto &s : init) {
    s.key = OBX_KEY_EMPTY;
    for (uint32_t a = 0; a < ph.n_aggs; a++) {
      if (ph.aggs[a].kind == OBX_AGG_MIN)
        s.cells[a][0] = (unsigned long long)INT64_MAX;
      else if (ph.aggs[a].kind == OBX_AGG_MAX)
        s.cells[a][0] = (unsigned long long)INT64_MIN;
    }
  }
  HIP_TRY(hipMemcpyAsync(h.d_gtable, init.data(),
                         sizeof(gslot) * OBX_GTABLE_SLOTS,
                         hipMemcpyHostToDevice, ctx->stream));
  HIP_TRY(hipMemsetAsync(h.d_counters, 0, 16 * 8, ctx->stream));

  /* Default: the fused single-kernel path (measured fastest in round 1).
     OBX_PIPELINE_AGG=1 switches to the filter->group->agg-pass kernel DAG
     (cleaner structure, currently slower — round-2 exploration; both paths
     are parity-tested). */
  const bool use_pipeline = getenv("OBX_PIPELINE_AGG") != nullptr;
  if (use_pipeline) {
    uint64_t bm_words = (h.total_rows + 63) / 64 + 1;
    if (!h.d_bitmap) HIP_TRY(hipMalloc(&h.d_bitmap, bm_words * 8));
    HIP_TRY(hipMemsetAsync(h.d_bitmap, 0, bm_words * 8, ctx->stream));
    if (!h.d_row_slot) HIP_TRY(hipMalloc(&h.d_row_slot, h.total_rows + 1));

    HIP_TRY(hipEventRecord(ctx->ev_start, ctx->stream));
    {
      auto kfn = h.lds_ok ? (ph.n_prog ? k_filter_prog_lds : k_filter_lds)
                          : (ph.n_prog ? k_filter_prog : k_filter);
      hipLaunchKernelGGL(kfn, dim3(grid_for(h.n_blocks)), dim3(OBX_WG_HOST), 0,
                         ctx->stream, h.d_buf, h.d_blocks, h.n_blocks,
                         h.d_pleaves, h.d_bleaves, ph, h.d_bitmap,
                         (int32_t *)nullptr, (uint32_t *)nullptr,
                         h.d_counters);
    }
    hipLaunchKernelGGL(k_group_pass, dim3(grid_for(h.n_blocks)),
                       dim3(OBX_WG_HOST), 0, ctx->stream, h.d_buf, h.d_blocks,
                       h.n_blocks, ph, h.d_bitmap, h.d_row_slot, h.d_gtable,
                       h.d_counters);
    for (uint32_t a = 0; a < ph.n_aggs;) {
      const dev_agg &ag = ph.aggs[a];
      if (ag.kind == OBX_AGG_COUNT && ag.ia == 0xFF) { a++; continue; }
      bool fuse = (ag.kind == OBX_AGG_SUM_PROD2 && a + 1 < ph.n_aggs &&
                   ph.aggs[a + 1].kind == OBX_AGG_SUM_PROD3 &&
                   ph.aggs[a + 1].ia == ag.ia && ph.aggs[a + 1].ib == ag.ib);
      hipLaunchKernelGGL(k_agg_pass, dim3(grid_for(h.n_blocks)),
                         dim3(OBX_WG_HOST), 0, ctx->stream, h.d_buf,
                         h.d_blocks, h.n_blocks, ph, a, h.d_bitmap,
                         h.d_row_slot, h.d_gtable);
      a += fuse ? 2 : 1;
    }
  } else {
    /* plan-specialized JIT kernel when eligible (compiled once per plan
       signature, before the timed region; obx_jit.inc) */
    jit_entry *je = jit_prepare(h, ph, plv);
    ctx->last_jit = je ? g_jit_kind : 0; /* 2 = persistent v2, 1 = v1 */
    HIP_TRY(hipEventRecord(ctx->ev_start, ctx->stream));
    if (je) {
      if (jit_launch(je, h, grid_for(h.n_blocks), ctx->stream) != 0)
        return OBX_INTERNAL_ERROR;
    } else {
      auto kfn = h.lds_ok
                     ? (ph.n_prog ? k_scan_filter_agg_prog_lds
                                  : k_scan_filter_agg_lds)
                     : (ph.n_prog ? k_scan_filter_agg_prog : k_scan_filter_agg);
      hipLaunchKernelGGL(kfn, dim3(grid_for(h.n_blocks)), dim3(OBX_WG_HOST), 0,
                         ctx->stream, h.d_buf, h.d_blocks, h.n_blocks,
                         h.d_pleaves, h.d_bleaves, ph, h.d_gtable,
                         h.d_counters);
    }
  }
  HIP_TRY(hipEventRecord(ctx->ev_stop, ctx->stream));
  HIP_TRY(hipStreamSynchronize(ctx->stream));
  float ms = 0;
  HIP_TRY(hipEventElapsedTime(&ms, ctx->ev_start, ctx->ev_stop));
  ctx->last_ms = ms;
  HIP_TRY(hipEventElapsedTime(&ms, ctx->ev_p0, ctx->ev_p1));
  ctx->last_prep_ms = ms;

  std::vector<gslot> gt(OBX_GTABLE_SLOTS);
  unsigned long long cnt[16];
  HIP_TRY(hipMemcpy(gt.data(), h.d_gtable, sizeof(gslot) * OBX_GTABLE_SLOTS,
                    hipMemcpyDeviceToHost));
  HIP_TRY(hipMemcpy(cnt, h.d_counters, 16 * 8, hipMemcpyDeviceToHost));
  if (cnt[1] != 0) {
    /* a workgroup shard overflowed its LDS group table: rerun with the
       direct-global high-cardinality kernel (growth path; AND-only
       filters — OR-programs keep the capacity error) */
    if (ph.n_prog != 0) return OBX_BUF_NOT_ENOUGH;
    if (!h.d_gtable_big)
      HIP_TRY(hipMalloc(&h.d_gtable_big, sizeof(gslot) * OBX_GTABLE_BIG));
    std::vector<gslot> init(OBX_GTABLE_BIG);
    memset(init.data(), 0, sizeof(gslot) * OBX_GTABLE_BIG);
    for (auto &s2 : init) {
      s2.key = OBX_KEY_EMPTY;
      for (uint32_t a = 0; a < ph.n_aggs; a++) {
        if (ph.aggs[a].kind == OBX_AGG_MIN)
          s2.cells[a][0] = (unsigned long long)INT64_MAX;
        else if (ph.aggs[a].kind == OBX_AGG_MAX)
          s2.cells[a][0] = (unsigned long long)INT64_MIN;
      }
    }
    HIP_TRY(hipMemcpyAsync(h.d_gtable_big, init.data(),
                           sizeof(gslot) * OBX_GTABLE_BIG,
                           hipMemcpyHostToDevice, ctx->stream));
    HIP_TRY(hipMemsetAsync(h.d_counters, 0, 16 * 8, ctx->stream));
    hipLaunchKernelGGL(k_scan_agg_direct, dim3(grid_for(h.n_blocks)),
                       dim3(OBX_WG_HOST), 0, ctx->stream, h.d_buf,
                       h.d_blocks, h.n_blocks, h.d_pleaves, h.d_bleaves,
                       ph, h.d_gtable_big, h.d_counters);
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    gt.resize(OBX_GTABLE_BIG);
    HIP_TRY(hipMemcpy(gt.data(), h.d_gtable_big,
                      sizeof(gslot) * OBX_GTABLE_BIG,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(cnt, h.d_counters, 16 * 8, hipMemcpyDeviceToHost));
    if (cnt[2] != 0) return OBX_BUF_NOT_ENOUGH; /* >4096 distinct groups */
  }
  /* slots 8..15: striped per-wave survivor sums */
  for (int i2 = 8; i2 < 16; i2++) cnt[0] += cnt[i2];

  memset(out, 0, sizeof(*out));
  out->rows_scanned = h.total_rows;
  out->rows_passed = cnt[0];
  uint8_t key_len = 0;
  if (agg)
    for (int g = 0; g < agg->n_group_cols; g++)
      key_len += h.cols[agg->group_cols[g]].len;
  std::vector<const gslot *> live;
  for (auto &s : gt)
    if (s.key != OBX_KEY_EMPTY) live.push_back(&s);
  std::sort(live.begin(), live.end(), [](const gslot *a, const gslot *b) {
    /* little-endian packed keys -> compare as memcmp on key bytes */
    uint8_t ka[8], kb[8];
    memcpy(ka, &a->key, 8);
    memcpy(kb, &b->key, 8);
    return memcmp(ka, kb, 8) < 0;
  });
  /* materialize ALL rows (the growth path's pagination source), then
     surface the inline result when it fits */
  h.last_rows.assign(live.size(), obx_group_row());
  for (size_t i = 0; i < live.size(); i++) {
    obx_group_row *g = &h.last_rows[i];
    memcpy(g->key, &live[i]->key, 8);
    g->key_len = key_len;
    g->row_count = live[i]->count;
    uint32_t na = agg ? agg->n_aggs : 0;
    for (uint32_t a = 0; a < na; a++) {
      uint8_t kind = agg->aggs[a].kind;
      if (kind == OBX_AGG_COUNT && agg->aggs[a].col_a == UINT16_MAX) {
        /* COUNT(*) = the group's row count (group pass) */
        g->cells[a].limb[0] = live[i]->count;
        g->cells[a].limb[1] = g->cells[a].limb[2] = g->cells[a].limb[3] = 0;
      } else if (kind == OBX_AGG_MIN || kind == OBX_AGG_MAX) {
        /* sign-extend the int64 min/max into the 256-bit cell; a group
           with NO non-null values keeps the init sentinel and its valid
           flag (cells[a][1]) unset — report 0, matching the oracle's
           empty-MIN/MAX convention (SQL NULL surfaces via row_count /
           COUNT(col) at the caller) */
        int64_t v = live[i]->cells[a][1] ? (int64_t)live[i]->cells[a][0] : 0;
        g->cells[a].limb[0] = (uint64_t)v;
        uint64_t s = v < 0 ? ~0ull : 0ull;
        g->cells[a].limb[1] = g->cells[a].limb[2] = g->cells[a].limb[3] = s;
      } else {
        for (int l = 0; l < 4; l++)
          g->cells[a].limb[l] = live[i]->cells[a][l];
      }
    }
  }
  if (h.last_rows.size() > OBX_MAX_GROUPS) {
    /* more groups than the inline result holds: rows stay paged behind
       obx_gpu_agg_fetch (the ObHashGroupByOp growth path) */
    out->n_groups = (uint32_t)h.last_rows.size();
    return OBX_BUF_NOT_ENOUGH;
  }
  out->n_groups = (uint32_t)h.last_rows.size();
  for (size_t i = 0; i < h.last_rows.size(); i++)
    out->groups[i] = h.last_rows[i];
  return OBX_SUCCESS;
}

extern "C" uint64_t obx_crc32c(const uint8_t *buf, int64_t len) {
  return host_crc32c(buf, len);
}

/* ---- CS load glue (obx_cs_load.cpp drives; these own the handle) ------- */
int obx_cs_alloc_handle(obx_gpu_ctx *ctx, obx_handle **h_out) {
  if (!ctx) return OBX_INVALID_ARGUMENT;
  HIP_TRY(hipSetDevice(ctx->device));
  *h_out = new obx_handle();
  return OBX_SUCCESS;
}

int obx_cs_finish_load(obx_gpu_ctx *ctx, obx_handle *h,
                       const obx_col_schema *cols, uint16_t n_cols,
                       std::vector<dev_block> &blocks, uint8_t *d_buf,
                       uint64_t arena_bytes, uint64_t total_rows) {
  h->d_buf = d_buf;
  h->n_blocks = (uint32_t)blocks.size();
  h->n_cols = n_cols;
  h->total_rows = total_rows;
  h->total_bytes = arena_bytes;
  memcpy(h->cols, cols, sizeof(obx_col_schema) * n_cols);
  bool lds_ok = true;
  for (uint32_t b = 0; b < h->n_blocks; b++) {
    const dev_block &db = blocks[b];
    if ((db.block_byte & 15) || db.block_len + 24 > OBX_LDS_STAGE_BYTES ||
        db.row_count > 4096)
      lds_ok = false;
    for (uint16_t c = 0; c < n_cols; c++) {
      const dev_col &dc = db.cols[c];
      uint8_t cls;
      switch (dc.enc) {
        case OBX_D_RAW: cls = 0; break;
        case OBX_D_DICT: cls = 1; break;
        case OBX_D_INTDIFF: cls = 2; break;
        case OBX_D_CONST: cls = dc.runs == 0 ? 3 : 4; break;
        default: cls = 4; break;
      }
      if (dc.flags & OBX_DF_STRING) cls |= 0x10;
      h->col_class.push_back(cls);
      h->col_cnt.push_back((uint8_t)(dc.count > 254 ? 255 : dc.count));
      if (b == 0) {
        h->col_dict_every[c] = true;
        h->col_dict_any[c] = true;
        h->col_raw8_every[c] = true;
        h->col_rangefam_every[c] = true;
        h->col_maxcnt[c] = 0;
      }
      if ((cls & 0xF) != 1 || (cls & 0x10) || dc.count > 63)
        h->col_dict_every[c] = false;
      if ((cls & 0xF) != 1 || dc.count > 63) h->col_dict_any[c] = false;
      if (!(dc.enc == OBX_D_RAW && !(dc.flags & OBX_DF_BITPACK) &&
            !(dc.flags & OBX_DF_STRING) && !(dc.flags & OBX_DF_HAS_EXT) &&
            dc.width == 8 && (dc.data_bit & 63) == 0))
        h->col_raw8_every[c] = false;
      if (!((dc.enc == OBX_D_RAW || dc.enc == OBX_D_INTDIFF) &&
            !(dc.flags & OBX_DF_STRING) && !(dc.flags & OBX_DF_HAS_EXT)))
        h->col_rangefam_every[c] = false;
      {
        uint32_t w = 255;
        if (dc.enc == OBX_D_RAW || dc.enc == OBX_D_DICT ||
            dc.enc == OBX_D_INTDIFF)
          w = (dc.flags & OBX_DF_BITPACK) ? dc.width
                                          : (uint32_t)dc.width * 8;
        if (w > h->col_maxw[c]) h->col_maxw[c] = w;
      }
      if (dc.flags & OBX_DF_HAS_EXT) h->col_ext_any[c] = true;
    }
    if (db.row_count > h->max_block_rows) h->max_block_rows = db.row_count;
    if (db.block_len > h->max_block_len) h->max_block_len = db.block_len;
  }
  h->lds_ok = lds_ok;
  /* dict payloads live only on the device for CS handles; the persistent
     JIT path requires the host-verified byte-identical dicts, so it is
     simply not taken (col_dict_stable stays false -> v1 JIT / generic
     kernels, which are parity-equal) */
  HIP_TRY(hipMalloc(&h->d_blocks, sizeof(dev_block) * blocks.size()));
  HIP_TRY(hipMemcpy(h->d_blocks, blocks.data(),
                    sizeof(dev_block) * blocks.size(),
                    hipMemcpyHostToDevice));
  HIP_TRY(hipMalloc(&h->d_counters, 16 * sizeof(unsigned long long)));
  HIP_TRY(hipMalloc(&h->d_gtable, sizeof(gslot) * OBX_GTABLE_SLOTS));
  HIP_TRY(hipMalloc(&h->d_pleaves, sizeof(dev_leaf) * OBX_DEV_MAX_LEAVES));
  {
    uint64_t nmm = (uint64_t)h->n_blocks * n_cols * 2;
    HIP_TRY(hipMalloc(&h->d_minmax, nmm * sizeof(int64_t)));
    hipLaunchKernelGGL(k_col_minmax, dim3(grid_for(h->n_blocks)),
                       dim3(OBX_WG_HOST), 0, nullptr, h->d_buf, h->d_blocks,
                       h->n_blocks, (uint32_t)n_cols, h->d_minmax);
    std::vector<int64_t> mm(nmm);
    HIP_TRY(hipMemcpy(mm.data(), h->d_minmax, nmm * sizeof(int64_t),
                      hipMemcpyDeviceToHost));
    for (uint16_t c = 0; c < n_cols; c++) {
      int64_t gmin = INT64_MAX, gmax = INT64_MIN;
      bool known = true;
      for (uint32_t b = 0; b < h->n_blocks; b++) {
        int64_t mn = mm[2 * ((uint64_t)b * n_cols + c)];
        int64_t mx = mm[2 * ((uint64_t)b * n_cols + c) + 1];
        if (mn == INT64_MIN && mx == INT64_MAX) { known = false; break; }
        if (mn <= mx) {
          if (mn < gmin) gmin = mn;
          if (mx > gmax) gmax = mx;
        }
      }
      h->col_known[c] = known;
      h->col_min[c] = gmin;
      h->col_max[c] = gmax;
    }
  }
  return OBX_SUCCESS;
}

int obx_cs_publish_handle(obx_gpu_ctx *ctx, obx_handle *h) {
  h->in_use = true;
  ctx->handles.push_back(*h);
  delete h;
  return (int)ctx->handles.size() - 1;
}

/* Debug/test-only: generate the hipRTC source the JIT would compile for a
 * plan against synthetic load-time column summaries, without a GPU (the
 * container has no device; tests hipcc-compile the dump offline).
 * col_flags bits: 1 = dict-everywhere, 2 = ext-any, 4 = bounds known.
 * Returns the source length (>= 0) or a negative status; 0 = the plan
 * would not take the JIT path. */
extern "C" int64_t obx_jit_dump_src(
    const obx_filter_desc *filter, const obx_agg_desc *agg,
    const obx_col_schema *cols, uint16_t n_cols, const uint8_t *col_flags,
    const int64_t *col_min, const int64_t *col_max,
    const uint32_t *col_maxcnt, const uint32_t *col_maxw,
    uint32_t max_block_rows, int force_v1,
    char *out, int64_t cap) {
  if (!cols || n_cols > OBX_DEV_MAX_COLS) return OBX_INVALID_ARGUMENT;
  obx_handle h;
  h.n_cols = n_cols;
  h.n_blocks = 1;
  h.lds_ok = true;
  h.max_block_rows = max_block_rows ? max_block_rows : 4096;
  h.max_block_len = OBX_LDS_STAGE_BYTES - 32;
  memcpy(h.cols, cols, sizeof(obx_col_schema) * n_cols);
  for (uint16_t c = 0; c < n_cols; c++) {
    h.col_dict_every[c] = (col_flags[c] & 1) != 0;
    h.col_ext_any[c] = (col_flags[c] & 2) != 0;
    h.col_known[c] = (col_flags[c] & 4) != 0;
    h.col_raw8_every[c] = (col_flags[c] & 8) != 0;
    h.col_rangefam_every[c] = (col_flags[c] & 16) != 0;
    h.col_dict_any[c] = h.col_dict_every[c] ||
        (obx_store_class(cols[c].obj_type) == OBX_SC_STRING &&
         (col_flags[c] & 1));
    h.col_dict_stable[c] = (col_flags[c] & 32) != 0 && h.col_dict_any[c];
    h.col_min[c] = col_min[c];
    h.col_max[c] = col_max[c];
    h.col_maxcnt[c] = col_maxcnt[c];
    h.col_maxw[c] = col_maxw ? col_maxw[c] : 255;
    uint8_t cls = h.col_dict_every[c] ? 1 : 0;
    if (obx_store_class(cols[c].obj_type) == OBX_SC_STRING) cls |= 0x10;
    h.col_class.push_back(cls);
    h.col_cnt.push_back(
        (uint8_t)(h.col_maxcnt[c] > 254 ? 255 : h.col_maxcnt[c]));
  }
  dev_plan_hdr ph;
  dev_leaf pl[OBX_DEV_MAX_LEAVES];
  int rc = build_plan(h, filter, agg, ph, pl);
  if (rc != OBX_SUCCESS) return rc;
  if (!agg) { /* bitmap-filter path: dump the wave-per-block filter JIT */
    jit_strategy fst;
    if (!jit_build_fstrategy(h, ph, pl, fst)) return 0;
    if (getenv("OBX_DUMP_RID") && fst.fstage) { /* test hook */
      fst.frid = 1;
      fst.fnch = (uint16_t)((h.max_block_rows + 63) / 64);
    }
    std::string fsrc = jit_gen_source_filter(ph, fst);
    if (out && cap > (int64_t)fsrc.size()) {
      memcpy(out, fsrc.data(), fsrc.size());
      out[fsrc.size()] = 0;
    }
    return (int64_t)fsrc.size();
  }
  jit_shape js;
  if (!jit_plan_shape(ph, js)) return 0;
  if (!jit_blocks_ok(h, ph, js)) return 0;
  jit_strategy st;
  if (!force_v1) jit_build_strategy(h, ph, js, pl, st);
  if (!st.ok && js.has_mm) return 0; /* v1 cannot do MIN/MAX */
  std::string src = st.ok ? jit_gen_source_v2(ph, js, st)
                          : jit_gen_source(ph, js);
  if (out && cap > (int64_t)src.size()) {
    memcpy(out, src.data(), src.size());
    out[src.size()] = 0;
  }
  return (int64_t)src.size();
}

extern "C" double obx_gpu_last_kernel_ms(obx_gpu_ctx *ctx) {
  return ctx ? ctx->last_ms : -1.0;
}

/* per-query monitoring (the rebuild of the reference's per-operator
 * sql_plan_monitor / NG_TRACE stats): device times of the
 * last query's prep (plan upload + k_lower_leaves) and main kernel. */
extern "C" double obx_gpu_last_prep_ms(obx_gpu_ctx *ctx) {
  return ctx ? ctx->last_prep_ms : -1.0;
}

extern "C" int obx_gpu_last_jit(obx_gpu_ctx *ctx) {
  return ctx ? ctx->last_jit : 0;
}

extern "C" uint64_t obx_gpu_total_rows(obx_gpu_ctx *ctx, int handle) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size()) return 0;
  return ctx->handles[handle].total_rows;
}
extern "C" uint64_t obx_gpu_total_bytes(obx_gpu_ctx *ctx, int handle) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size()) return 0;
  return ctx->handles[handle].total_bytes;
}
extern "C" uint64_t obx_gpu_last_survivors(obx_gpu_ctx *ctx, int handle) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size()) return 0;
  return ctx->handles[handle].last_survivors;
}

/* paged access to the last scan's full (sorted) group rows — the result
 * surface for > OBX_MAX_GROUPS groups (the reference's hash group-by
 * grows unboundedly, ob_exec_hash_struct_vec.h:1718; our device table
 * holds OBX_GTABLE_SLOTS groups and the host pages them out). */
extern "C" int obx_gpu_agg_fetch(obx_gpu_ctx *ctx, int handle,
                                 uint32_t start, uint32_t count,
                                 obx_group_row *out, uint32_t *n_out,
                                 uint64_t *n_total) {
  if (!ctx || handle < 0 || handle >= (int)ctx->handles.size() || !out)
    return OBX_INVALID_ARGUMENT;
  obx_handle &h = ctx->handles[handle];
  uint64_t total = h.last_rows.size();
  if (n_total) *n_total = total;
  uint32_t n = 0;
  for (; n < count && start + n < total; n++) out[n] = h.last_rows[start + n];
  if (n_out) *n_out = n;
  return OBX_SUCCESS;
}
