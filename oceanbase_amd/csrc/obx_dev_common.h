/*
 * obx_dev_common.h — device-side helpers for the OLAP hot-path kernels.
 * Shared between the precompiled kernels (obx_kernels.hip) and the
 * hipRTC-JIT specialized scan kernels (obx_jit.cpp): bit readers, column
 * decode contexts, filter-leaf evaluation, exact int128 arithmetic, the
 * LDS group table and LDS block staging. Compiled under hipRTC the HIP
 * runtime header and <cstdint> are unavailable, so the types come from
 * obx_dev.h's hiprtc branch.
 */
#ifndef OBX_DEV_COMMON_H
#define OBX_DEV_COMMON_H
#if !defined(__HIPRTC__) && !defined(OBX_HIPRTC)
#include <hip/hip_runtime.h>
#endif
#include "obx_dev.h"


#define WG 256
#define WAVES (WG / 64)
#ifndef OBX_PIPELINE
#define OBX_PIPELINE 0   /* 0 = single-buffer staging (higher occupancy; measured faster), 1 = double-buffered prefetch */
#endif

/* view of a block's bytes: either the global buffer (bias 0) or the LDS
 * staging copy (bias = block_byte * 8). All dev_col offsets are absolute;
 * reads subtract the bias. */
struct blk_view {
  const uint8_t *base;
  uint64_t bit_bias;   /* slow paths: absolute offsets minus this bias */
  uint64_t rbase_bit;  /* ctx fast paths: block-relative offsets plus this */
};

/* ---------------- bit reads (base is >=16-B aligned) -------------------- */
__device__ __forceinline__ uint64_t bit_read_at(const uint8_t *__restrict__ b,
                                                uint64_t bitpos, uint32_t k) {
  const uint64_t *w = (const uint64_t *)b;
  uint64_t widx = bitpos >> 6;
  uint32_t sh = (uint32_t)(bitpos & 63);
  uint64_t v = w[widx] >> sh;
  if (sh + k > 64) v |= w[widx + 1] << (64 - sh);
  if (k < 64) v &= (((uint64_t)1 << k) - 1);
  return v;
}

__device__ __forceinline__ uint64_t bit_read(const blk_view &bv,
                                             uint64_t bitpos, uint32_t k) {
  return bit_read_at(bv.base, bitpos - bv.bit_bias, k);
}

__device__ __forceinline__ int64_t sext(uint64_t v, uint32_t bytes) {
  if (bytes >= 8) return (int64_t)v;
  uint32_t sh = 64 - bytes * 8;
  return ((int64_t)(v << sh)) >> sh;
}

/* byte-lexicographic order mapping for char values (== the oracle's
 * char_key: low `len` LE bytes -> big-endian integer) */
__device__ __forceinline__ int64_t dev_char_key(int64_t raw_le, uint32_t len) {
  return (int64_t)(__builtin_bswap64((uint64_t)raw_le) >> (8 * (8 - len)));
}

__device__ __forceinline__ uint64_t shfl64(uint64_t v, int lane) {
  uint32_t lo = __shfl((int)(uint32_t)v, lane, 64);
  uint32_t hi = __shfl((int)(uint32_t)(v >> 32), lane, 64);
  return ((uint64_t)hi << 32) | lo;
}
__device__ __forceinline__ uint64_t shflxor64(uint64_t v, int mask) {
  uint32_t lo = __shfl_xor((int)(uint32_t)v, mask, 64);
  uint32_t hi = __shfl_xor((int)(uint32_t)(v >> 32), mask, 64);
  return ((uint64_t)hi << 32) | lo;
}

/* unpack the column's packed stream entry for row r (refs/diffs/values) */
__device__ __forceinline__ uint64_t col_packed(const blk_view &bv,
                                               const dev_col &c, uint32_t r) {
  uint32_t W = (c.flags & OBX_DF_BITPACK) ? c.width : (uint32_t)c.width * 8;
  return bit_read(bv, c.data_bit + (uint64_t)r * W, W);
}

__device__ __forceinline__ bool col_is_null_ext(const blk_view &bv,
                                                const dev_col &c, uint32_t r) {
  if (!(c.flags & OBX_DF_HAS_EXT)) return false;
  return bit_read(bv, c.ext_bit + (uint64_t)r * c.ext_width, c.ext_width) != 0;
}

/* RLE: ref of row r by binary search over run starts
 * (ob_rle_decoder.cpp:18-31) */
__device__ __forceinline__ uint64_t rle_ref(const blk_view &bv,
                                            const dev_col &c, uint32_t r) {
  uint64_t lo = 0, hi = c.runs;
  while (lo < hi) {
    uint64_t mid = (lo + hi) >> 1;
    uint64_t s = bit_read(bv, (c.aux_byte + mid * c.rib) * 8,
                          (uint32_t)c.rib * 8);
    if (s <= r) lo = mid + 1; else hi = mid;
  }
  uint64_t run = lo - 1;
  return bit_read(bv,
                  (c.aux_byte + (uint64_t)c.runs * c.rib + run * c.rfb) * 8,
                  (uint32_t)c.rfb * 8);
}

__device__ __forceinline__ int64_t dict_entry(const blk_view &bv,
                                              const dev_col &c, uint64_t ref) {
  uint64_t v = bit_read(bv, (c.dict_byte + ref * c.entry_len) * 8,
                        (uint32_t)c.entry_len * 8);
  if (c.flags & OBX_DF_STRING) return (int64_t)v; /* raw LE bytes */
  return (c.flags & OBX_DF_SIGNED) ? sext(v, c.tss) : sext(v, c.entry_len);
}

/* ---------------- per-block register contexts ---------------------------
 * Built ONCE per block from the (uniform but vector-loaded) descriptors so
 * the row loops touch only registers; without this the compiler re-issues
 * global byte loads of descriptor fields per row (SMEM scalarization is
 * blocked by the kernel's global stores). */
struct col_ctx {
  /* offsets are BLOCK-RELATIVE bits (blocks are <= 17 KB, so 32 bits
     suffice) — halves the uniform (SGPR) footprint of a context; the
     102-SGPR budget was spilling to scratch inside the row loops */
  uint32_t data_bit;   /* packed stream start (block-relative bits) */
  uint32_t aux;        /* DICT: dict payload bit pos (block-relative) */
  uint32_t ext_bit;    /* ext-bit stream (HAS_EXT, block-relative) */
  uint32_t count;      /* DICT: count (null ref >= count); else 0 */
  int64_t base;        /* INTDIFF/CONST value */
  uint16_t W;          /* packed entry width in bits */
  uint8_t entry_bits;  /* DICT: dict entry width in bits */
  uint8_t sext_sh;     /* value: shift for sign extension (0 = none) */
  uint8_t ent_sh;      /* DICT entry sign-extension shift */
  uint8_t kind;        /* 0 raw, 1 dict, 2 intdiff, 3 const, 4 slow(RLE/exc) */
  uint8_t has_ext;
  uint8_t ext_w;
};

__device__ __forceinline__ col_ctx make_col_ctx(const dev_col &c,
                                                uint64_t blk_bit) {
  col_ctx x;
  x.data_bit = (uint32_t)(c.data_bit - blk_bit);
  x.aux = 0; x.base = 0;
  x.ext_bit = (uint32_t)(c.ext_bit - blk_bit);
  x.count = 0;
  x.W = (c.flags & OBX_DF_BITPACK) ? c.width : (uint16_t)(c.width * 8);
  x.entry_bits = 0; x.sext_sh = 0; x.ent_sh = 0;
  x.has_ext = (c.flags & OBX_DF_HAS_EXT) ? 1 : 0;
  x.ext_w = c.ext_width;
  switch (c.enc) {
    case OBX_D_RAW:
      x.kind = 0;
      if (!(c.flags & OBX_DF_BITPACK) && !(c.flags & OBX_DF_STRING))
        x.sext_sh = (uint8_t)(64 - 8 * ((c.flags & OBX_DF_SIGNED) ? c.tss
                                                                  : c.width));
      if (x.sext_sh == 0 || x.sext_sh >= 64) x.sext_sh = 0;
      /* aligned 8-byte raw (Q1's l_extendedprice shape): one aligned
         64-bit load replaces the generic bit-funnel read */
      if (x.W == 64 && x.sext_sh == 0 && ((blk_bit + x.data_bit) & 63) == 0)
        x.kind = 5;
      break;
    case OBX_D_DICT:
      x.kind = 1;
      x.count = c.count;
      x.aux = (uint32_t)(c.dict_byte * 8 - blk_bit);
      x.entry_bits = (uint8_t)(c.entry_len * 8);
      if (!(c.flags & OBX_DF_STRING)) {
        uint32_t sb = (c.flags & OBX_DF_SIGNED) ? c.tss : c.entry_len;
        x.ent_sh = (uint8_t)(sb >= 8 ? 0 : 64 - 8 * sb);
      }
      break;
    case OBX_D_INTDIFF:
      x.kind = 2;
      x.base = c.base;
      break;
    case OBX_D_CONST:
      if (c.runs == 0) {
        x.kind = 3;
        x.base = c.base;
        x.count = c.count; /* 0 = null const */
      } else {
        x.kind = 4;
      }
      break;
    default:
      x.kind = 4; /* RLE and exotic: slow path */
      break;
  }
  return x;
}

__device__ __forceinline__ int64_t ctx_value(const blk_view &bv,
                                             const col_ctx &x, uint32_t r,
                                             bool &null_out) {
  null_out = false;
  switch (x.kind) {
    case 0: { /* RAW */
      if (x.has_ext && bit_read_at(bv.base,
                                   bv.rbase_bit + x.ext_bit +
                                       (uint64_t)r * x.ext_w, x.ext_w)) {
        null_out = true; return 0;
      }
      uint64_t v = bit_read_at(bv.base,
                               bv.rbase_bit + x.data_bit + (uint64_t)r * x.W,
                               x.W);
      return x.sext_sh ? (((int64_t)(v << x.sext_sh)) >> x.sext_sh)
                       : (int64_t)v;
    }
    case 1: { /* DICT */
      uint64_t ref = bit_read_at(
          bv.base, bv.rbase_bit + x.data_bit + (uint64_t)r * x.W, x.W);
      if (ref >= x.count) { null_out = true; return 0; }
      uint64_t v = bit_read_at(bv.base,
                               bv.rbase_bit + x.aux + ref * x.entry_bits,
                               x.entry_bits);
      return x.ent_sh ? (((int64_t)(v << x.ent_sh)) >> x.ent_sh) : (int64_t)v;
    }
    case 2: { /* INTDIFF */
      if (x.has_ext && bit_read_at(bv.base,
                                   bv.rbase_bit + x.ext_bit +
                                       (uint64_t)r * x.ext_w, x.ext_w)) {
        null_out = true; return 0;
      }
      return (int64_t)((uint64_t)x.base +
                       bit_read_at(bv.base,
                                   bv.rbase_bit + x.data_bit +
                                       (uint64_t)r * x.W, x.W));
    }
    case 3: { /* CONST, no exceptions */
      if (x.count == 0) { null_out = true; return 0; }
      return x.base;
    }
    case 5: { /* aligned raw64 */
      if (x.has_ext && bit_read_at(bv.base,
                                   bv.rbase_bit + x.ext_bit +
                                       (uint64_t)r * x.ext_w, x.ext_w)) {
        null_out = true; return 0;
      }
      const uint64_t byte0 = (bv.rbase_bit + x.data_bit) >> 3;
      return *(const int64_t *)(bv.base + byte0 + ((uint64_t)r << 3));
    }
    default:
      return 0; /* caller uses the slow path for kind 4 */
  }
}

/* per-block filter-leaf context (registers) */
struct leaf_ctx {
  col_ctx cc;              /* decode context of the filter column */
  uint64_t mask, lo, hi;   /* REF_MASK / packed RANGE (mask = xor bias) */
  uint8_t mode, invert, op, slow;
};

__device__ __forceinline__ leaf_ctx make_leaf_ctx(const dev_block &blk,
                                                  const dev_leaf &plf,
                                                  const blk_leaf &blf,
                                                  uint64_t blk_bit) {
  const dev_col &c = blk.cols[plf.col];
  leaf_ctx x;
  x.cc = make_col_ctx(c, blk_bit);
  x.mask = blf.mask; x.lo = blf.lo; x.hi = blf.hi;
  x.mode = blf.mode; x.invert = blf.invert; x.op = plf.op;
  /* slow fallback: RLE/exception decode, IN lists, or black programs
     (operand arrays stay in the global leaf) */
  x.slow = (x.cc.kind == 4 ||
            (blf.mode == OBX_LEAF_VALUE &&
             (plf.op == 7 || plf.op == 10))) ? 1 : 0;
  return x;
}

__device__ __forceinline__ bool leaf_ctx_match(const blk_view &bv,
                                               const leaf_ctx &x,
                                               const dev_leaf &plf,
                                               uint32_t r) {
  switch (x.mode) {
    case OBX_LEAF_NONE: return false;
    case OBX_LEAF_ALL:
      if (x.cc.has_ext &&
          bit_read_at(bv.base, bv.rbase_bit + x.cc.ext_bit +
                                   (uint64_t)r * x.cc.ext_w, x.cc.ext_w))
        return false;
      return true;
    case OBX_LEAF_REF_MASK: {
      uint64_t ref = bit_read_at(
          bv.base, bv.rbase_bit + x.cc.data_bit + (uint64_t)r * x.cc.W,
          x.cc.W);
      return (x.mask >> ref) & 1;
    }
    case OBX_LEAF_RANGE: {
      if (x.cc.has_ext &&
          bit_read_at(bv.base, bv.rbase_bit + x.cc.ext_bit +
                                   (uint64_t)r * x.cc.ext_w, x.cc.ext_w))
        return false;
      uint64_t v = bit_read_at(bv.base, bv.rbase_bit + x.cc.data_bit +
                                            (uint64_t)r * x.cc.W, x.cc.W)
                   ^ x.mask; /* signed domains are order-mapped by xor bias */
      bool in = (v - x.lo) <= (x.hi - x.lo);
      return in != (bool)x.invert;
    }
    case OBX_LEAF_NULL: {
      bool isn;
      (void)ctx_value(bv, x.cc, r, isn);
      return isn != (bool)x.invert;
    }
    default: { /* VALUE (op != IN); operands read from the plan leaf */
      bool isn;
      int64_t v = ctx_value(bv, x.cc, r, isn);
      if (x.op == 8) return isn;
      if (x.op == 9) return !isn;
      if (isn) return false;
      if (plf.char_len) v = dev_char_key(v, plf.char_len);
      switch (x.op) {
        case 0: return v == plf.vlo;
        case 1: return v <= plf.vlo;
        case 2: return v < plf.vlo;
        case 3: return v >= plf.vlo;
        case 4: return v > plf.vlo;
        case 5: return v != plf.vlo;
        case 6: return v >= plf.vlo && v <= plf.vhi;
      }
      return false;
    }
  }
}

/* full value decode: int64 (sign-extended), is_null out.
 * __noinline__: slow/cold generic path — keeps register pressure off the
 * hot ctx_value loops. */
/* HEX_PACKING / STRING_DIFF row decode (ObHexStringUnpacker /
 * ObStringDiffHeader::copy_string). __noinline__ with unrolling disabled:
 * this is a cold generic path, and its register footprint rolls up into
 * every kernel that can reach col_value (AMDGPU call-graph budget). */
__device__ __noinline__ int64_t str_value(const blk_view bv,
                                          const dev_col &c, uint32_t r) {
  uint64_t v = 0;
  const uint64_t row_byte = c.data_bit / 8 + (uint64_t)r * c.width;
  if (c.enc == OBX_D_HEX) {
#pragma clang loop unroll(disable)
    for (uint32_t i = 0; i < c.datum_len; i++) {
      uint8_t b8 = (uint8_t)bit_read(bv, (row_byte + i / 2) * 8, 8);
      uint8_t nib = (uint8_t)((b8 >> (((i + 1) % 2) * 4)) & 0xF);
      uint8_t ch8 = (uint8_t)bit_read(bv, (c.dict_byte + nib) * 8, 8);
      v |= (uint64_t)ch8 << (8 * i);
    }
    return (int64_t)v;
  }
  if (c.enc == OBX_D_SPREFIX) {
    /* prefix ref nibble + suffix (ObStringPrefixCellHeader semantics for
       fixed-length cells); prefix lengths via the cumulative end index */
    const bool hexp = c.entry_len > 0;
    uint32_t ref = (uint32_t)bit_read(bv, row_byte * 8, 8) & 0xF;
    if (ref >= c.count) return 0;
    uint64_t pstart = 0, pend = 0;
    if (ref > 0)
      pstart = bit_read(bv, (c.aux_byte + (uint64_t)(ref - 1) * c.rib) * 8,
                        (uint32_t)c.rib * 8);
    pend = bit_read(bv, (c.aux_byte + (uint64_t)ref * c.rib) * 8,
                    (uint32_t)c.rib * 8);
    uint32_t plen = (uint32_t)(pend - pstart);
    uint64_t pdata = c.aux_byte + (uint64_t)c.count * c.rib;
#pragma clang loop unroll(disable)
    for (uint32_t i = 0; i < plen; i++)
      v |= bit_read(bv, (pdata + pstart + i) * 8, 8) << (8 * i);
#pragma clang loop unroll(disable)
    for (uint32_t i = 0; i < c.datum_len - plen; i++) {
      uint64_t ch8;
      if (hexp) {
        uint64_t b8 = bit_read(bv, (row_byte + 1 + i / 2) * 8, 8);
        ch8 = bit_read(
            bv, (c.dict_byte + ((b8 >> (((i + 1) % 2) * 4)) & 0xF)) * 8, 8);
      } else {
        ch8 = bit_read(bv, (row_byte + 1 + i) * 8, 8);
      }
      v |= ch8 << (8 * (plen + i));
    }
    return (int64_t)v;
  }
  /* STRING_DIFF */
  const bool hex = c.entry_len > 0;
  const uint64_t chars_byte = c.dict_byte + c.runs;
  uint32_t pos = 0, cpos = 0, dpos = 0;
#pragma clang loop unroll(disable)
  for (uint32_t i = 0; i < c.runs; i++) {
    uint8_t d8 = (uint8_t)bit_read(bv, (c.dict_byte + i) * 8, 8);
    uint32_t cnt = d8 >> 1;
#pragma clang loop unroll(disable)
    for (uint32_t j = 0; j < cnt; j++, pos++) {
      uint8_t ch8;
      if (d8 & 1) {
        if (hex) {
          uint8_t b8 = (uint8_t)bit_read(bv, (row_byte + dpos / 2) * 8, 8);
          uint8_t nib = (uint8_t)((b8 >> (((dpos + 1) % 2) * 4)) & 0xF);
          ch8 = (uint8_t)bit_read(bv, (chars_byte + nib) * 8, 8);
        } else {
          ch8 = (uint8_t)bit_read(bv, (row_byte + dpos) * 8, 8);
        }
        dpos++;
      } else {
        ch8 = (uint8_t)bit_read(bv, (c.aux_byte + cpos) * 8, 8);
        cpos++;
      }
      v |= (uint64_t)ch8 << (8 * pos);
    }
  }
  return (int64_t)v;
}

__device__ __noinline__ int64_t col_value(const blk_view bv,
                                             const dev_col &c, uint32_t r,
                                             bool &null_out) {
  null_out = false;
  switch (c.enc) {
    case OBX_D_RAW: {
      if (col_is_null_ext(bv, c, r)) { null_out = true; return 0; }
      uint64_t v = col_packed(bv, c, r);
      if (c.flags & OBX_DF_BITPACK) return (int64_t)v; /* zero-extended */
      if (c.flags & OBX_DF_STRING) return (int64_t)v;  /* raw LE bytes */
      return (c.flags & OBX_DF_SIGNED) ? sext(v, c.tss)
                                       : sext(v, c.width);
    }
    case OBX_D_DICT: {
      uint64_t ref = col_packed(bv, c, r);
      if (ref >= c.count) { null_out = true; return 0; }
      return dict_entry(bv, c, ref);
    }
    case OBX_D_RLE: {
      uint64_t ref = rle_ref(bv, c, r);
      if (ref >= c.count) { null_out = true; return 0; }
      return dict_entry(bv, c, ref);
    }
    case OBX_D_CONST: {
      if (c.runs == 0) {
        if (c.count == 0) { null_out = true; return 0; }
        return c.base;
      }
      uint64_t ref = c.rfb; /* const_ref */
      for (uint32_t i = 0; i < c.runs; i++) {
        uint64_t rid = bit_read(
            bv, (c.aux_byte + c.runs + (uint64_t)i * c.rib) * 8,
            (uint32_t)c.rib * 8);
        if (rid == r) {
          ref = bit_read(bv, (c.aux_byte + i) * 8, 8);
          break;
        }
      }
      if (ref >= c.count) { null_out = true; return 0; }
      return dict_entry(bv, c, ref);
    }
    case OBX_D_INTDIFF: {
      if (col_is_null_ext(bv, c, r)) { null_out = true; return 0; }
      uint64_t diff = col_packed(bv, c, r);
      return (int64_t)((uint64_t)c.base + diff);
    }
    case OBX_D_HEX:
    case OBX_D_SDIFF:
    case OBX_D_SPREFIX: {
      if (col_is_null_ext(bv, c, r)) { null_out = true; return 0; }
      return str_value(bv, c, r);
    }
  }
  return 0;
}

/* generic VALUE-mode leaf (operands order-mapped by the host for char) */
/* col_value with block context: handles COLUMN_EQUAL (exception rows,
 * else the reference column's value — one level deep by construction,
 * ob_column_equal_encoder.h; cold generic path). */
__device__ __noinline__ int64_t col_value2(const blk_view bv,
                                           const dev_block &blk,
                                           const dev_col &c, uint32_t r,
                                           bool &null_out) {
  if (c.enc == OBX_D_EQUAL) {
    null_out = false;
#pragma clang loop unroll(disable)
    for (uint32_t i = 0; i < c.runs; i++) {
      uint64_t rid = bit_read(bv, (c.dict_byte + (uint64_t)i * c.rib) * 8,
                              (uint32_t)c.rib * 8);
      if (rid == r) {
        if (bit_read(bv, c.aux_byte * 8 + i, 1)) {
          null_out = true;
          return 0;
        }
        uint64_t v = bit_read(
            bv,
            (c.aux_byte + (c.runs + 7) / 8 + (uint64_t)i * c.datum_len) * 8,
            (uint32_t)c.datum_len * 8);
        if (c.flags & OBX_DF_STRING) return (int64_t)v;
        return sext(v, (c.flags & OBX_DF_SIGNED) ? c.tss : c.datum_len);
      }
      if (rid > r) break;
    }
    return col_value(bv, blk.cols[c.width], r, null_out); /* width=ref col */
  }
  if (c.enc == OBX_D_SUBSTR) {
    null_out = false;
#pragma clang loop unroll(disable)
    for (uint32_t i = 0; i < c.runs; i++) {
      uint64_t rid = bit_read(bv, (c.dict_byte + (uint64_t)i * c.rib) * 8,
                              (uint32_t)c.rib * 8);
      if (rid == r) {
        if (bit_read(bv, c.aux_byte * 8 + i, 1)) {
          null_out = true;
          return 0;
        }
        return (int64_t)bit_read(
            bv,
            (c.aux_byte + (c.runs + 7) / 8 + (uint64_t)i * c.datum_len) * 8,
            (uint32_t)c.datum_len * 8);
      }
      if (rid > r) break;
    }
    int64_t rv = col_value(bv, blk.cols[c.width], r, null_out);
    if (null_out) return 0;
    uint64_t mask2 =
        c.datum_len >= 8 ? ~0ull : ((1ull << (8 * c.datum_len)) - 1);
    return (int64_t)(((uint64_t)rv >> (8 * (uint32_t)c.base)) & mask2);
  }
  return col_value(bv, c, r, null_out);
}

/* Black (generic-expression) filter: device evaluation of the postfix
 * OBX_BX_* program with three-valued logic — independent of the oracle's
 * C implementation, compared bit-exactly by the parity tests.
 * __noinline__: cold generic path (register budget). */
__device__ __noinline__ bool black_eval_vals(const dev_leaf &plf,
                                             const int64_t *vals,
                                             const bool *nus) {
  int64_t st[8];
  bool nu[8];
  int sp = 0;
#pragma clang loop unroll(disable)
  for (int p = 0; p < plf.n_bprog; p++) {
    uint8_t op = plf.bprog[p];
    if (op < 0x40) {
      if (op >= plf.n_bcols || sp >= 8) return false;
      st[sp] = vals[op];
      nu[sp++] = nus[op];
    } else if (op < 0x50) {
      int k = op & 0x0F;
      if (k >= 4 || sp >= 8) return false;
      st[sp] = plf.bconst[k];
      nu[sp++] = false;
    } else if (op == 0x54) {
      if (sp < 1) return false;
      st[sp - 1] = (int64_t)(0 - (uint64_t)st[sp - 1]);
    } else if (op == 0x72) {
      if (sp < 1) return false;
      if (!nu[sp - 1]) st[sp - 1] = !st[sp - 1];
    } else {
      if (sp < 2) return false;
      sp--;
      int64_t a = st[sp - 1], b2 = st[sp];
      bool na = nu[sp - 1], nb = nu[sp];
      bool rn = na || nb;
      int64_t r2 = 0;
      switch (op) {
        case 0x50: r2 = (int64_t)((uint64_t)a + (uint64_t)b2); break;
        case 0x51: r2 = (int64_t)((uint64_t)a - (uint64_t)b2); break;
        case 0x52: r2 = (int64_t)((uint64_t)a * (uint64_t)b2); break;
        case 0x53:
          if (b2 == 0) rn = true;
          else if (a == INT64_MIN && b2 == -1) r2 = a;
          else r2 = a / b2;
          break;
        case 0x55: /* MOD: x % 0 -> NULL; INT64_MIN % -1 == 0 */
          if (b2 == 0) rn = true;
          else if (a == INT64_MIN && b2 == -1) r2 = 0;
          else r2 = a % b2;
          break;
        case 0x60: r2 = a < b2; break;
        case 0x61: r2 = a <= b2; break;
        case 0x62: r2 = a > b2; break;
        case 0x63: r2 = a >= b2; break;
        case 0x64: r2 = a == b2; break;
        case 0x65: r2 = a != b2; break;
        case 0x70:
          if ((!na && !a) || (!nb && !b2)) { r2 = 0; rn = false; }
          else if (rn) r2 = 0;
          else r2 = 1;
          break;
        case 0x71:
          if ((!na && a) || (!nb && b2)) { r2 = 1; rn = false; }
          else if (rn) r2 = 0;
          else r2 = 0;
          break;
        default: return false;
      }
      st[sp - 1] = r2;
      nu[sp - 1] = rn;
    }
  }
  if (sp != 1) return false;
  return !nu[0] && st[0] != 0;
}

__device__ __noinline__ bool black_match(const blk_view bv,
                                         const dev_block &blk,
                                         const dev_leaf &plf, uint32_t r) {
  int64_t vals[4];
  bool nus[4];
#pragma clang loop unroll(disable)
  for (int j = 0; j < plf.n_bcols; j++)
    vals[j] = col_value2(bv, blk, blk.cols[plf.bcols[j]], r, nus[j]);
  return black_eval_vals(plf, vals, nus);
}

__device__ __forceinline__ bool leaf_value_match(const dev_leaf &lf, int64_t v,
                                                 bool isn) {
  if (lf.op == 8) return isn;
  if (lf.op == 9) return !isn;
  if (isn) return false;
  if (lf.char_len) v = dev_char_key(v, lf.char_len);
  switch (lf.op) {
    case 0: return v == lf.vlo;
    case 1: return v <= lf.vlo;
    case 2: return v < lf.vlo;
    case 3: return v >= lf.vlo;
    case 4: return v > lf.vlo;
    case 5: return v != lf.vlo;
    case 6: return v >= lf.vlo && v <= lf.vhi;
    case 7: {
      for (int i = 0; i < lf.n_in; i++)
        if (v == lf.in_list[i]) return true;
      return false;
    }
  }
  return false;
}

__device__ __noinline__ bool leaf_match(const blk_view bv,
                                           const dev_block &blk,
                                           const dev_leaf &plf,
                                           const blk_leaf &blf, uint32_t r) {
  const dev_col &c = blk.cols[plf.col];
  switch (blf.mode) {
    case OBX_LEAF_NONE: return false;
    case OBX_LEAF_ALL:
      if (c.enc == OBX_D_RAW || c.enc == OBX_D_INTDIFF)
        return !col_is_null_ext(bv, c, r);
      return true;
    case OBX_LEAF_REF_MASK: {
      uint64_t ref = (c.enc == OBX_D_RLE) ? rle_ref(bv, c, r)
                                          : col_packed(bv, c, r);
      return (blf.mask >> ref) & 1; /* host guarantees count<=63 here */
    }
    case OBX_LEAF_RANGE: {
      if (col_is_null_ext(bv, c, r)) return false;
      uint64_t v = col_packed(bv, c, r) ^ blf.mask;
      bool in = (v - blf.lo) <= (blf.hi - blf.lo);
      return in != (bool)blf.invert;
    }
    case OBX_LEAF_NULL: {
      bool isn;
      (void)col_value2(bv, blk, c, r, isn);
      return isn != (bool)blf.invert;
    }
    default: {
      if (plf.op == 10) return black_match(bv, blk, plf, r);
      bool isn;
      int64_t v = col_value2(bv, blk, c, r, isn);
      return leaf_value_match(plf, v, isn);
    }
  }
}

/* ---------------- 128-bit helpers --------------------------------------- */
struct i128v { uint64_t lo, hi; };
__device__ __forceinline__ i128v i128_from_i64(int64_t v) {
  i128v r; r.lo = (uint64_t)v; r.hi = (uint64_t)(v >> 63); return r;
}
__device__ __forceinline__ i128v i128_mul_i64(int64_t a, int64_t b) {
  __int128 p = (__int128)a * (__int128)b;
  i128v r; r.lo = (uint64_t)p; r.hi = (uint64_t)((unsigned __int128)p >> 64);
  return r;
}
__device__ __forceinline__ i128v i128_mul_pos_i64(i128v a, int64_t m) {
  /* (int128 a) * (non-negative int64 m); result must fit int128
     (bounded for decimal p<=18 inputs — DESIGN.md) */
  unsigned __int128 av = ((unsigned __int128)a.hi << 64) | a.lo;
  unsigned __int128 p = av * (unsigned __int128)(uint64_t)m;
  i128v r; r.lo = (uint64_t)p; r.hi = (uint64_t)(p >> 64);
  return r;
}

/* wave-wide sum of an i128 (every lane receives the total) */
__device__ __forceinline__ i128v wave_sum_i128(i128v v) {
  for (int off = 32; off > 0; off >>= 1) {
    uint64_t olo = shflxor64(v.lo, off);
    uint64_t ohi = shflxor64(v.hi, off);
    uint64_t lo = v.lo + olo;
    v.hi = v.hi + ohi + (lo < olo);
    v.lo = lo;
  }
  return v;
}

/* ---------------- LDS group table ---------------------------------------
 * Cells are 4-way lane-striped (stripe = lane & 3): same-slot atomic adds
 * from one wave serialize only within a 16-lane stripe group, cutting the
 * measured LDS same-address conflict cycles ~4x. Stripes merge at flush. */
#ifndef OBX_STRIPES
#define OBX_STRIPES 2
#endif
struct lds_table {
  unsigned long long key[OBX_LTABLE_SLOTS];
  unsigned long long count[OBX_LTABLE_SLOTS][OBX_STRIPES];
  unsigned long long cell[OBX_LTABLE_SLOTS][OBX_DEV_MAX_AGGS][OBX_STRIPES][2];
};

__device__ __forceinline__ uint32_t key_hash(uint64_t k) {
  k *= 0x9E3779B97F4A7C15ull;
  return (uint32_t)(k >> 59) & (OBX_LTABLE_SLOTS - 1);
}

__device__ __forceinline__ int lds_slot(lds_table *t, uint64_t key) {
  uint32_t idx = key_hash(key);
  for (int probe = 0; probe < OBX_LTABLE_SLOTS; probe++) {
    unsigned long long k = t->key[idx];      /* fast path: plain read */
    if (k == key) return (int)idx;
    if (k == OBX_KEY_EMPTY) {
      unsigned long long cur = atomicCAS(&t->key[idx], OBX_KEY_EMPTY,
                                         (unsigned long long)key);
      if (cur == OBX_KEY_EMPTY || cur == key) return (int)idx;
    }
    idx = (idx + 1) & (OBX_LTABLE_SLOTS - 1);
  }
  return -1;
}

__device__ __forceinline__ void lds_acc_i128(unsigned long long *cell,
                                             i128v v) {
  unsigned long long old = atomicAdd(&cell[0], (unsigned long long)v.lo);
  uint64_t carry = ((uint64_t)old + v.lo < v.lo) ? 1 : 0;
  unsigned long long hi = (unsigned long long)(v.hi + carry);
  if (hi) atomicAdd(&cell[1], hi);
}

/* CAS-based signed min/max on a u64 holding an int64 (LDS or global) */
__device__ __forceinline__ void cas_minmax(unsigned long long *slot,
                                           int64_t v, bool is_min) {
  unsigned long long cur = *slot;
  for (;;) {
    int64_t c = (int64_t)cur;
    if (is_min ? (v >= c) : (v <= c)) return;
    unsigned long long prev = atomicCAS(slot, cur, (unsigned long long)v);
    if (prev == cur) return;
    cur = prev;
  }
}

/* 256-bit global accumulate of a signed 128-bit value */
__device__ __forceinline__ void g_acc_i128(unsigned long long *limbs,
                                           uint64_t lo, uint64_t hi) {
  uint64_t v[4];
  v[0] = lo; v[1] = hi;
  v[2] = v[3] = ((int64_t)hi < 0) ? ~0ull : 0ull;
  uint64_t c = 0;
  for (int i = 0; i < 4; i++) {
    uint64_t add = v[i] + c;
    uint64_t c1 = (add < c) ? 1 : 0;
    if (add == 0) { c = c1; continue; }
    unsigned long long old = atomicAdd(&limbs[i], (unsigned long long)add);
    c = c1 + (((uint64_t)old + add) < add ? 1 : 0);
  }
}

/* issue one block's bytes as async LDS-DMA (global_load_lds_dwordx4;
 * block_byte is 16-B aligned, container blocks are 16-B aligned with zero
 * padding). Completion: the issuing thread's s_waitcnt vmcnt(0), then a
 * workgroup barrier. */
__device__ __forceinline__ void stage_issue(const uint8_t *__restrict__ buf,
                                            const dev_block &blk,
                                            uint8_t *lds_blk) {
  const uint32_t n16 = (blk.block_len + 8 + 15) >> 4;
  const uint8_t *src = buf + blk.block_byte;
  for (uint32_t i = threadIdx.x; i < n16; i += WG) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t *)(src +
                                                             (size_t)i * 16),
        (__attribute__((address_space(3))) uint32_t *)(lds_blk +
                                                       (size_t)i * 16),
        16, 0, 0);
  }
}

__device__ __forceinline__ void stage_wait() {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
}

/* full group-key build (cold path: slow group columns or first occurrence
 * of a dict-ref cell). __noinline__ keeps its register use off the hot
 * loops. NULL group values set flag bits in the top key byte. */
__device__ __noinline__ uint64_t build_group_key(
    const blk_view bv, uint32_t n_group_cols, const dev_col *gd0,
    const dev_col *gd1, uint32_t kl0, uint32_t kl1, uint32_t r) {
  uint64_t key = 0;
  if (n_group_cols > 0) {
    bool isn;
    int64_t kv = col_value(bv, *gd0, r, isn);
    if (isn) key |= 1ull << 56;
    else key |= (uint64_t)kv &
                ((kl0 >= 8) ? ~0ull : (((uint64_t)1 << (kl0 * 8)) - 1));
    if (n_group_cols > 1) {
      bool isn1;
      int64_t kv1 = col_value(bv, *gd1, r, isn1);
      uint64_t kb = 0;
      if (isn1) key |= 1ull << 57;
      else kb = (uint64_t)kv1 &
                ((kl1 >= 8) ? ~0ull : (((uint64_t)1 << (kl1 * 8)) - 1));
      key |= kb << (kl0 * 8);
    }
  }
  return key;
}

/* ---------------- fused scan->filter->aggregate kernel ------------------ */
#ifdef OBX_FAST_LEAVES_OVR
#define OBX_FAST_LEAVES OBX_FAST_LEAVES_OVR
#else
#define OBX_FAST_LEAVES 4
#endif
#define OBX_FAST_NEED 8
#define OBX_MAX_BLOCK_ROWS 2048

/* three-valued fold of the filter combine program over per-block leaf
 * classes (0 = never passes, 1 = always passes, 2 = row-dependent);
 * implements the executor tree's constant-result short-circuit
 * (ob_pushdown_filter.cpp:1559-1632) at block granularity. */
__device__ __forceinline__ uint8_t leaf_class(const dev_block &cur,
                                              const dev_leaf &plf,
                                              const blk_leaf &blf) {
  if (blf.mode == OBX_LEAF_NONE) return 0;
  if (blf.mode == OBX_LEAF_ALL &&
      !(cur.cols[plf.col].flags & OBX_DF_HAS_EXT))
    return 1;
  return 2;
}

__device__ __forceinline__ uint8_t fold_prog3(
    const dev_plan_hdr &ph, const dev_block &cur,
    const dev_leaf *__restrict__ pl, const blk_leaf *__restrict__ bl) {
  if (ph.n_leaves == 0) return 1;
  if (ph.n_prog == 0) { /* AND of all leaves */
    uint8_t res = 1;
    for (uint32_t i = 0; i < ph.n_leaves; i++) {
      uint8_t c = leaf_class(cur, pl[i], bl[i]);
      if (c == 0) return 0;
      if (c == 2) res = 2;
    }
    return res;
  }
  uint8_t stack[16];
  int sp = 0;
  for (uint32_t p = 0; p < ph.n_prog; p++) {
    uint8_t t = ph.prog[p];
    if (t < ph.n_leaves) {
      stack[sp++] = leaf_class(cur, pl[t], bl[t]);
    } else {
      uint8_t b2 = stack[--sp], a2 = stack[sp - 1];
      if (t == 128) /* AND */
        stack[sp - 1] = (a2 == 0 || b2 == 0) ? 0
                        : (a2 == 1 && b2 == 1) ? 1 : 2;
      else /* OR */
        stack[sp - 1] = (a2 == 1 || b2 == 1) ? 1
                        : (a2 == 0 && b2 == 0) ? 0 : 2;
    }
  }
  return stack[0];
}

typedef __attribute__((address_space(3))) uint64_t lds3_u64;
#endif /* OBX_DEV_COMMON_H */
