/* CS column/block layer restatement — see obx_cs_block.h for the cited
 * reference functions. Oracle/test infrastructure only. */
#include "obx_cs_block.h"

#include <stdlib.h>
#include <string.h>

static const uint32_t WB[4] = {1, 2, 4, 8};

static inline int in_null(const uint8_t *nulls, uint32_t r) {
  return nulls && ((nulls[r >> 3] >> (r & 7)) & 1);
}

/* get_byte_packed_int_size: smallest of 1/2/4/8 covering v */
static inline uint32_t byte_packed(uint64_t v) {
  return v <= 0xFF ? 1 : v <= 0xFFFF ? 2 : v <= 0xFFFFFFFFull ? 4 : 8;
}

static inline uint8_t width_tag(uint32_t wb) {
  return wb == 1 ? 0 : wb == 2 ? 1 : wb == 4 ? 2 : 3;
}

/* ObCSEncodingUtil::get_bit_size: bits needed to store v */
static inline uint32_t bit_size(uint64_t v) {
  uint32_t b = 1;
  while (v >>= 1) b++;
  return b;
}

static int64_t codec_enc(uint8_t t, const uint8_t *in, uint32_t n,
                         uint32_t wb, uint8_t *out, size_t cap) {
  switch (t) {
    case OBX_CS_ENC_DELTA_ZIGZAG_RLE: return obx_cs_dzr_enc(in, n, wb, out, cap);
    case OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_RLE: return obx_cs_ddzr_enc(in, n, wb, out, cap);
    case OBX_CS_ENC_DELTA_ZIGZAG_PFOR: return obx_cs_dzp_enc(in, n, wb, out, cap);
    case OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_PFOR: return obx_cs_ddzp_enc(in, n, wb, out, cap);
    case OBX_CS_ENC_SIMD_FIXEDPFOR: return obx_cs_fpfor_enc(in, n, wb, out, cap);
    case OBX_CS_ENC_XOR_FIXED_PFOR: return obx_cs_xpfor_enc(in, n, wb, out, cap);
    default: return -1;
  }
}

static int64_t codec_dec(uint8_t t, const uint8_t *in, size_t len,
                         uint32_t n, uint32_t wb, uint8_t *out) {
  switch (t) {
    case OBX_CS_ENC_DELTA_ZIGZAG_RLE: return obx_cs_dzr_dec(in, len, n, wb, out);
    case OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_RLE: return obx_cs_ddzr_dec(in, len, n, wb, out);
    case OBX_CS_ENC_DELTA_ZIGZAG_PFOR: return obx_cs_dzp_dec(in, len, n, wb, out);
    case OBX_CS_ENC_DOUBLE_DELTA_ZIGZAG_PFOR: return obx_cs_ddzp_dec(in, len, n, wb, out);
    case OBX_CS_ENC_SIMD_FIXEDPFOR: return obx_cs_fpfor_dec(in, len, n, wb, out);
    case OBX_CS_ENC_XOR_FIXED_PFOR: return obx_cs_xpfor_dec(in, len, n, wb, out);
    default: return -1;
  }
}

/* uint stream with caller-fixed width: used for offset arrays
 * (build_offset_array_stream_meta, ob_stream_encoding_struct.cpp:
 * 182-202) and dict ref streams (build_unsigned_stream_meta with
 * min 0 -> no base; the width covers ref_stream_max_value_, which can
 * exceed the array's own max, ob_dict_column_encoder.cpp:159-181) */
static int64_t uint_stream_enc(const uint64_t *vals, uint32_t n,
                               uint8_t enc_type, uint32_t wb,
                               uint8_t *buf, size_t cap) {
  if (!n) return -1;
  obx_cs_int_meta m;
  memset(&m, 0, sizeof(m));
  m.version = 1;
  m.type = enc_type ? enc_type : OBX_CS_ENC_RAW;
  m.width_tag = width_tag(wb);
  int hn = obx_cs_int_meta_enc(&m, buf, cap);
  if (hn < 0) return -1;
  size_t pos = (size_t)hn;
  uint8_t *packed = (uint8_t *)malloc((size_t)n * wb);
  if (!packed) return -1;
  for (uint32_t i = 0; i < n; i++)
    memcpy(packed + (size_t)i * wb, &vals[i], wb);
  int64_t dn;
  if (m.type == OBX_CS_ENC_RAW) {
    dn = (int64_t)n * wb;
    if (pos + (size_t)dn > cap) { free(packed); return -1; }
    memcpy(buf + pos, packed, (size_t)dn);
  } else {
    dn = codec_enc(m.type, packed, n, wb, buf + pos, cap - pos);
  }
  free(packed);
  if (dn < 0) return -1;
  return (int64_t)pos + dn;
}

static int64_t offset_stream_enc(const uint64_t *ends, uint32_t n,
                                 uint8_t enc_type, uint8_t *buf,
                                 size_t cap) {
  if (!n) return -1;
  return uint_stream_enc(ends, n, enc_type, byte_packed(ends[n - 1]), buf,
                         cap);
}

static int64_t offset_stream_dec(const uint8_t *buf, size_t len, uint32_t n,
                                 uint64_t *ends) {
  obx_cs_int_meta m;
  int hn = obx_cs_int_meta_dec(buf, len, &m);
  if (hn < 0) return -1;
  uint32_t wb = WB[m.width_tag];
  size_t pos = (size_t)hn;
  uint8_t *packed = (uint8_t *)malloc((size_t)n * wb);
  if (!packed) return -1;
  int64_t dn;
  if (m.type == OBX_CS_ENC_RAW) {
    dn = (int64_t)n * wb;
    if (pos + (size_t)dn > len) { free(packed); return -1; }
    memcpy(packed, buf + pos, (size_t)dn);
  } else {
    dn = codec_dec(m.type, buf + pos, len - pos, n, wb, packed);
  }
  if (dn < 0) { free(packed); return -1; }
  for (uint32_t i = 0; i < n; i++) {
    uint64_t v = 0;
    memcpy(&v, packed + (size_t)i * wb, wb);
    ends[i] = v;
  }
  free(packed);
  return (int64_t)pos + dn;
}

int64_t obx_cs_block_enc(uint32_t rows, uint32_t ncols,
                         const obx_cs_col_in *cols, uint8_t *buf,
                         size_t cap) {
  if (!rows || !ncols || ncols > OBX_CS_MAX_COLS || !cols || !buf)
    return -1;
  const size_t hdr_sz = sizeof(obx_cs_block_header);
  const size_t heads = hdr_sz + sizeof(obx_cs_all_col_header) +
                       (size_t)ncols * sizeof(obx_cs_col_header);
  const uint32_t bitmap_size = (rows + 7) / 8;
  if (heads + 64 > cap) return -1;

  obx_cs_col_header chdr[OBX_CS_MAX_COLS];
  memset(chdr, 0, sizeof(chdr));
  uint64_t soff[OBX_CS_MAX_STREAMS];
  uint32_t n_streams = 0;

  /* pooled string bytes accumulate here, appended after the columns
   * (store_all_string_data_, ob_micro_block_cs_encoder.cpp:1247-1303;
   * NONE_COMPRESSOR path) */
  size_t str_pool_cap = 0;
  for (uint32_t c = 0; c < ncols; c++)
    if (cols[c].is_string)
      for (uint32_t r = 0; r < rows; r++)
        str_pool_cap += cols[c].lens ? cols[c].lens[r] : 0;
  str_pool_cap += (size_t)rows * 8 * ncols + 64; /* fixed-null padding */
  uint8_t *pool = (uint8_t *)malloc(str_pool_cap);
  if (!pool) return -1;
  size_t pool_len = 0;

  size_t pos = heads;
  int64_t rc = -1;
  for (uint32_t c = 0; c < ncols; c++) {
    const obx_cs_col_in *ci = &cols[c];
    uint32_t null_cnt = 0;
    for (uint32_t r = 0; r < rows; r++)
      if (in_null(ci->nulls, r)) null_cnt++;
    if (n_streams + 3 > OBX_CS_MAX_STREAMS) goto fail;

    if (ci->want_dict) {
      /* ---- INT_DICT / STR_DICT (ob_dict_column_encoder.cpp,
       * ob_int_dict_column_encoder.cpp, ob_str_dict_column_encoder.cpp):
       * [ObDictEncodingMeta][dict value stream(s)][ref stream] ---- */
      chdr[c].type = ci->is_string ? OBX_CS_COL_STR_DICT
                                   : OBX_CS_COL_INT_DICT;
      chdr[c].obj_type = ci->is_string ? OBX_OBJ_VARCHAR : OBX_OBJ_INT;
      uint32_t nn = rows - null_cnt;
      obx_cs_dict_meta dm;
      memset(&dm, 0, sizeof(dm));
      dm.attrs = OBX_CS_DICT_IS_SORTED; /* do_sort_dict_ always runs */
      if (null_cnt) dm.attrs |= OBX_CS_DICT_HAS_NULL;
      uint32_t *refs = (uint32_t *)malloc((size_t)rows * 4);
      uint32_t distinct = 0;
      int64_t *idv = NULL;
      const uint8_t **sps = NULL; /* distinct string ptrs */
      uint32_t *sls = NULL;       /* distinct string lens */
      if (!refs) goto fail;
      if (!ci->is_string) {
        if (!ci->ivals) { free(refs); goto fail; }
        idv = (int64_t *)malloc((size_t)(nn ? nn : 1) * 8);
        if (!idv) { free(refs); goto fail; }
        uint32_t k = 0;
        for (uint32_t r = 0; r < rows; r++)
          if (!in_null(ci->nulls, r)) idv[k++] = ci->ivals[r];
        /* sort + unique (ht sort_dict: ascending value order) */
        for (uint32_t i = 1; i < k; i++) { /* insertion sort is fine for
                                              oracle-scale blocks */
          int64_t x = idv[i];
          uint32_t j = i;
          while (j && idv[j - 1] > x) { idv[j] = idv[j - 1]; j--; }
          idv[j] = x;
        }
        for (uint32_t i = 0; i < k; i++)
          if (!distinct || idv[distinct - 1] != idv[i])
            idv[distinct++] = idv[i];
        for (uint32_t r = 0; r < rows; r++) {
          if (in_null(ci->nulls, r)) { refs[r] = distinct; continue; }
          uint32_t lo = 0, hi = distinct;
          while (lo + 1 < hi) {
            uint32_t mid = (lo + hi) / 2;
            if (idv[mid] <= ci->ivals[r]) lo = mid; else hi = mid;
          }
          refs[r] = lo;
        }
      } else {
        if (!ci->lens) { free(refs); goto fail; }
        const uint8_t **rp =
            (const uint8_t **)malloc((size_t)rows * sizeof(void *));
        sps = (const uint8_t **)malloc((size_t)rows * sizeof(void *) + 1);
        sls = (uint32_t *)malloc((size_t)rows * 4 + 4);
        if (!rp || !sps || !sls) { free(rp); goto fail_dict; }
        const uint8_t *src = ci->bytes;
        for (uint32_t r = 0; r < rows; r++) {
          rp[r] = NULL;
          if (!in_null(ci->nulls, r)) { rp[r] = src; src += ci->lens[r]; }
        }
        /* distinct, sorted by bytewise (memcmp, then length) order */
        for (uint32_t r = 0; r < rows; r++) {
          if (!rp[r]) continue;
          uint32_t l = ci->lens[r];
          uint32_t lo = 0, hi = distinct;
          int found = 0;
          while (lo < hi) {
            uint32_t mid = (lo + hi) / 2;
            uint32_t ml = sls[mid] < l ? sls[mid] : l;
            int cr = memcmp(sps[mid], rp[r], ml);
            if (!cr) cr = (sls[mid] > l) - (sls[mid] < l);
            if (cr == 0) { found = 1; lo = mid; break; }
            if (cr < 0) lo = mid + 1; else hi = mid;
          }
          if (!found) {
            memmove(sps + lo + 1, sps + lo,
                    (size_t)(distinct - lo) * sizeof(void *));
            memmove(sls + lo + 1, sls + lo, (size_t)(distinct - lo) * 4);
            sps[lo] = rp[r];
            sls[lo] = l;
            distinct++;
          }
        }
        for (uint32_t r = 0; r < rows; r++) {
          if (!rp[r]) { refs[r] = distinct; continue; }
          uint32_t l = ci->lens[r], lo = 0, hi = distinct;
          while (lo < hi) {
            uint32_t mid = (lo + hi) / 2;
            uint32_t ml = sls[mid] < l ? sls[mid] : l;
            int cr = memcmp(sps[mid], rp[r], ml);
            if (!cr) cr = (sls[mid] > l) - (sls[mid] < l);
            if (cr == 0) { lo = mid; break; }
            if (cr < 0) lo = mid + 1; else hi = mid;
          }
          refs[r] = lo;
        }
        free(rp);
      }
      dm.distinct_val_cnt = distinct;

      if (distinct == 0) {
        /* all rows null: only the dict meta is stored
         * (build_ref_encoder_ctx_ :70-76, store_dict_ref_ :225-227) */
        dm.ref_row_cnt = 0;
        if (pos + sizeof(dm) > cap) goto fail_dict;
        memcpy(buf + pos, &dm, sizeof(dm));
        pos += sizeof(dm);
        free(refs);
        free(idv);
        free(sps);
        free(sls);
        continue;
      }

      /* const-ref trial (try_const_encoding_ref_ :139-181) */
      uint64_t max_ref = distinct - 1 + (null_cnt ? 1 : 0);
      uint32_t *cnt = (uint32_t *)calloc(distinct + 1, 4);
      if (!cnt) goto fail_dict;
      for (uint32_t r = 0; r < rows; r++) cnt[refs[r]]++;
      uint32_t const_ref = 0, max_cnt = 0;
      for (uint32_t i = 0; i <= distinct; i++)
        if (cnt[i] > max_cnt) { max_cnt = cnt[i]; const_ref = i; }
      free(cnt);
      uint32_t ec = rows - max_cnt;
      int use_const = 0;
      uint64_t ref_max;
      if (ec == 0) {
        use_const = 1;
        dm.attrs |= OBX_CS_DICT_CONST_REF;
        dm.ref_row_cnt = 2;
        ref_max = const_ref;
      } else if (ec <= OBX_CS_DICT_MAX_EXCEPTIONS &&
                 ec < (uint64_t)rows * OBX_CS_DICT_MAX_EXCEPTION_PCT /
                          100) {
        use_const = 1;
        dm.attrs |= OBX_CS_DICT_CONST_REF;
        dm.ref_row_cnt = 2 + 2 * ec;
        uint32_t ex_max_row = 0;
        for (int64_t r = (int64_t)rows - 1; r >= 0; r--)
          if (refs[r] != const_ref) { ex_max_row = (uint32_t)r; break; }
        ref_max = ec;
        if (ex_max_row > ref_max) ref_max = ex_max_row;
        if (max_ref > ref_max) ref_max = max_ref;
      } else {
        dm.ref_row_cnt = rows;
        ref_max = max_ref;
      }
      if (pos + sizeof(dm) > cap) goto fail_dict;
      memcpy(buf + pos, &dm, sizeof(dm));
      pos += sizeof(dm);

      /* dict value stream(s) */
      if (!ci->is_string) {
        int64_t n = obx_cs_int_stream_enc3(
            idv, NULL, distinct, ci->enc_type ? ci->enc_type
                                              : OBX_CS_ENC_RAW,
            0, 0, buf + pos, cap - pos);
        if (n < 0) goto fail_dict;
        pos += (size_t)n;
        soff[n_streams++] = pos;
      } else {
        uint64_t total = 0;
        int fixed = 1;
        for (uint32_t i = 0; i < distinct; i++) {
          total += sls[i];
          if (sls[i] != sls[0]) fixed = 0;
        }
        obx_cs_str_meta sm;
        memset(&sm, 0, sizeof(sm));
        sm.uncompressed_len = (uint32_t)total;
        if (fixed) {
          sm.attr |= OBX_CS_STR_FIXED_LEN;
          sm.fixed_str_len = sls[0];
        }
        int hn = obx_cs_str_meta_enc(&sm, buf + pos, cap - pos);
        if (hn < 0) goto fail_dict;
        pos += (size_t)hn;
        soff[n_streams++] = pos;
        if (pool_len + total > str_pool_cap) goto fail_dict;
        for (uint32_t i = 0; i < distinct; i++) {
          memcpy(pool + pool_len, sps[i], sls[i]);
          pool_len += sls[i];
        }
        if (!fixed) {
          uint64_t *ends = (uint64_t *)malloc((size_t)distinct * 8);
          if (!ends) goto fail_dict;
          uint64_t acc = 0;
          for (uint32_t i = 0; i < distinct; i++) {
            acc += sls[i];
            ends[i] = acc;
          }
          int64_t n = offset_stream_enc(ends, distinct, ci->enc_type,
                                        buf + pos, cap - pos);
          free(ends);
          if (n < 0) goto fail_dict;
          pos += (size_t)n;
          soff[n_streams++] = pos;
        }
      }

      /* ref stream (do_store_dict_ref_, ob_dict_column_encoder.h:58-109;
       * const layout: [exception_cnt][const_ref][row ids][refs]) */
      {
        uint64_t *ra = (uint64_t *)malloc((size_t)dm.ref_row_cnt * 8);
        if (!ra) goto fail_dict;
        if (use_const) {
          ra[0] = ec;
          ra[1] = const_ref;
          uint32_t idx = 0;
          for (uint32_t r = 0; r < rows && idx < ec; r++)
            if (refs[r] != const_ref) {
              ra[2 + idx] = r;
              ra[2 + ec + idx] = refs[r];
              idx++;
            }
        } else {
          for (uint32_t r = 0; r < rows; r++) ra[r] = refs[r];
        }
        int64_t n = uint_stream_enc(ra, dm.ref_row_cnt, OBX_CS_ENC_RAW,
                                    byte_packed(ref_max), buf + pos,
                                    cap - pos);
        free(ra);
        if (n < 0) goto fail_dict;
        pos += (size_t)n;
        soff[n_streams++] = pos;
      }
      free(refs);
      free(idv);
      free(sps);
      free(sls);
      continue;
    fail_dict:
      free(refs);
      free(idv);
      free(sps);
      free(sls);
      goto fail;
    }

    if (!ci->is_string) {
      /* ---- INTEGER column (ob_integer_column_encoder.cpp) ---- */
      if (!ci->ivals) goto fail;
      int64_t mn = 0, mx = 0;
      int any = 0;
      for (uint32_t r = 0; r < rows; r++) {
        if (in_null(ci->nulls, r)) continue;
        if (!any || ci->ivals[r] < mn) mn = ci->ivals[r];
        if (!any || ci->ivals[r] > mx) mx = ci->ivals[r];
        any = 1;
      }
      if (!any) mn = mx = 0;
      /* null handling (:190-220): replace with a value adjacent to the
       * range; bitmap only when the range spans the column's whole
       * STORE type (INTEGER_MASK_TABLE[type_store_size_], :183-186) */
      uint32_t sw = ci->store_width ? ci->store_width : 8;
      int64_t ts_min =
          sw == 8 ? INT64_MIN : -((int64_t)1 << (8 * sw - 1));
      int64_t ts_max =
          sw == 8 ? INT64_MAX : ((int64_t)1 << (8 * sw - 1)) - 1;
      int use_replace = 0, use_bitmap = 0;
      int64_t replace = 0;
      if (!any && null_cnt > 0) {
        /* all datums null: build_signed_stream_meta(0, 0, replace, 0)
         * (ob_integer_column_encoder.cpp:79-89) */
        use_replace = 1;
        replace = 0;
      } else if (null_cnt > 0) {
        if (mn == 0) {
          if (mx != ts_max) { use_replace = 1; replace = mx + 1; }
          else { use_replace = 1; replace = -1; }
        } else if (mn == ts_min) {
          if (mx != ts_max) { use_replace = 1; replace = mx + 1; }
          else { use_bitmap = 1; }
        } else {
          use_replace = 1;
          replace = mn - 1;
        }
      }
      chdr[c].type = OBX_CS_COL_INTEGER;
      /* ObObjType per store width (ob_obj_type.h: TinyInt=1, SmallInt=2,
       * Int32=4, Int=5) */
      chdr[c].obj_type = sw == 1 ? 1 : sw == 2 ? 2 : sw == 4 ? 4
                                                 : OBX_OBJ_INT;
      if (use_bitmap) {
        chdr[c].attrs |= OBX_CS_CA_HAS_NULL_BITMAP;
        if (pos + bitmap_size > cap) goto fail;
        memset(buf + pos, 0, bitmap_size);
        for (uint32_t r = 0; r < rows; r++)
          if (in_null(ci->nulls, r))
            buf[pos + r / 8] |= (uint8_t)(1 << (7 - r % 8)); /* MSB-first */
        pos += bitmap_size;
      }
      int64_t n = obx_cs_int_stream_enc3(
          ci->ivals, null_cnt ? ci->nulls : NULL, rows,
          ci->enc_type ? ci->enc_type : OBX_CS_ENC_RAW, use_replace,
          replace, buf + pos, cap - pos);
      if (n < 0) goto fail;
      pos += (size_t)n;
      soff[n_streams++] = pos;
    } else {
      /* ---- STRING column (ob_string_column_encoder.cpp:53-135) ---- */
      if (!ci->lens || (!ci->bytes && str_pool_cap)) goto fail;
      int64_t fix_size = -1;
      int has_zero = 0, any = 0;
      uint64_t var_total = 0;
      for (uint32_t r = 0; r < rows; r++) {
        if (in_null(ci->nulls, r)) continue;
        uint32_t l = ci->lens[r];
        var_total += l;
        if (l == 0) has_zero = 1;
        if (!any) fix_size = l;
        else if (fix_size != (int64_t)l) fix_size = -2;
        any = 1;
      }
      if (fix_size < 0) fix_size = -1;
      int use_fixed = 0, use_zero_null = 0, use_bitmap = 0;
      if (null_cnt > 0) {
        if (has_zero) {
          use_bitmap = 1;
          use_fixed = fix_size >= 0;
        } else if (fix_size >= 0) {
          /* cost rule (:73-90, non-raw estimate) */
          uint64_t offs_est =
              ((uint64_t)bit_size((uint64_t)fix_size) + 1) * rows / 8;
          if ((uint64_t)fix_size * null_cnt + bitmap_size < offs_est) {
            use_fixed = 1;
            use_bitmap = 1;
          } else {
            use_zero_null = 1;
          }
        } else {
          use_zero_null = 1;
        }
      } else {
        use_fixed = fix_size >= 0;
      }
      chdr[c].type = OBX_CS_COL_STRING;
      chdr[c].obj_type = OBX_OBJ_VARCHAR;
      if (use_fixed) chdr[c].attrs |= OBX_CS_CA_IS_FIXED;
      if (use_bitmap) {
        chdr[c].attrs |= OBX_CS_CA_HAS_NULL_BITMAP;
        if (pos + bitmap_size > cap) goto fail;
        memset(buf + pos, 0, bitmap_size);
        for (uint32_t r = 0; r < rows; r++)
          if (in_null(ci->nulls, r))
            buf[pos + r / 8] |= (uint8_t)(1 << (7 - r % 8));
        pos += bitmap_size;
      }
      obx_cs_str_meta sm;
      memset(&sm, 0, sizeof(sm));
      if (use_zero_null) sm.attr |= OBX_CS_STR_ZERO_LEN_NULL;
      if (use_fixed) {
        sm.attr |= OBX_CS_STR_FIXED_LEN;
        sm.fixed_str_len = (uint32_t)(fix_size < 0 ? 0 : fix_size);
        sm.uncompressed_len = sm.fixed_str_len * rows;
      } else {
        sm.uncompressed_len = (uint32_t)var_total;
      }
      int hn = obx_cs_str_meta_enc(&sm, buf + pos, cap - pos);
      if (hn < 0) goto fail;
      pos += (size_t)hn;
      soff[n_streams++] = pos;
      /* bytes -> pool (do_convert_datum_to_stream_: fixed nulls are
       * zero-filled placeholders; var nulls contribute nothing) */
      const uint8_t *src = ci->bytes;
      uint64_t ends_buf_stack[1];
      (void)ends_buf_stack;
      if (use_fixed) {
        uint32_t fl = sm.fixed_str_len;
        if (pool_len + (size_t)fl * rows > str_pool_cap) goto fail;
        for (uint32_t r = 0; r < rows; r++) {
          if (in_null(ci->nulls, r)) {
            memset(pool + pool_len, 0, fl);
          } else {
            memcpy(pool + pool_len, src, fl);
            src += fl;
          }
          pool_len += fl;
        }
      } else {
        uint64_t *ends = (uint64_t *)malloc((size_t)rows * 8);
        if (!ends) goto fail;
        uint64_t acc = 0;
        for (uint32_t r = 0; r < rows; r++) {
          if (!in_null(ci->nulls, r)) {
            uint32_t l = ci->lens[r];
            if (pool_len + l > str_pool_cap) { free(ends); goto fail; }
            memcpy(pool + pool_len, src, l);
            src += l;
            pool_len += l;
            acc += l;
          }
          ends[r] = acc;
        }
        int64_t n = offset_stream_enc(ends, rows, ci->enc_type, buf + pos,
                                      cap - pos);
        free(ends);
        if (n < 0) goto fail;
        pos += (size_t)n;
        soff[n_streams++] = pos;
      }
    }
  }

  /* pooled string data */
  if (pos + pool_len > cap) goto fail;
  memcpy(buf + pos, pool, pool_len);
  pos += pool_len;

  /* block-tail stream-offset stream (store_stream_offsets_) */
  uint32_t sol = 0;
  if (n_streams) {
    int64_t n = offset_stream_enc(soff, n_streams, OBX_CS_ENC_RAW,
                                  buf + pos, cap - pos);
    if (n < 0) goto fail;
    sol = (uint32_t)n;
    pos += (size_t)n;
  }

  /* fill headers */
  {
    obx_cs_block_header bh;
    memset(&bh, 0, sizeof(bh));
    bh.magic = OBX_CS_BLOCK_MAGIC;
    bh.version = 1;
    bh.header_size = (uint16_t)hdr_sz;
    bh.row_count = rows;
    bh.column_count = (uint16_t)ncols;
    memcpy(buf, &bh, sizeof(bh));
    obx_cs_all_col_header ach;
    memset(&ach, 0, sizeof(ach));
    ach.all_string_data_length = (uint32_t)pool_len;
    ach.stream_offsets_length = sol;
    ach.stream_count = (uint16_t)n_streams;
    memcpy(buf + hdr_sz, &ach, sizeof(ach));
    memcpy(buf + hdr_sz + sizeof(ach), chdr,
           (size_t)ncols * sizeof(obx_cs_col_header));
  }
  rc = (int64_t)pos;
fail:
  free(pool);
  return rc;
}

int obx_cs_block_dec(const uint8_t *buf, size_t len,
                     obx_cs_block_view *v) {
  memset(v, 0, sizeof(*v));
  obx_cs_block_header bh;
  if (len < sizeof(bh)) return -1;
  memcpy(&bh, buf, sizeof(bh));
  if (bh.magic != OBX_CS_BLOCK_MAGIC || bh.version != 1 ||
      bh.column_count == 0 || bh.column_count > OBX_CS_MAX_COLS)
    return -1;
  size_t hp = bh.header_size;
  if (hp + sizeof(obx_cs_all_col_header) > len) return -1;
  memcpy(&v->ach, buf + hp, sizeof(v->ach));
  hp += sizeof(v->ach);
  if (hp + (size_t)bh.column_count * sizeof(obx_cs_col_header) > len)
    return -1;
  v->buf = buf;
  v->len = len;
  v->rows = bh.row_count;
  v->ncols = bh.column_count;
  for (uint32_t c = 0; c < v->ncols; c++)
    memcpy(&v->col[c].h, buf + hp + c * sizeof(obx_cs_col_header),
           sizeof(obx_cs_col_header));
  hp += (size_t)bh.column_count * sizeof(obx_cs_col_header);

  /* tail: [... pooled strings][stream-offset stream] */
  if (v->ach.stream_count > OBX_CS_MAX_STREAMS) return -1;
  if ((size_t)v->ach.stream_offsets_length +
          v->ach.all_string_data_length > len)
    return -1;
  const size_t so_start = len - v->ach.stream_offsets_length;
  v->all_string = buf + so_start - v->ach.all_string_data_length;
  v->stream_count = v->ach.stream_count;
  if (v->stream_count) {
    uint64_t ends[OBX_CS_MAX_STREAMS];
    if (offset_stream_dec(buf + so_start, v->ach.stream_offsets_length,
                          v->stream_count, ends) < 0)
      return -1;
    for (uint32_t i = 0; i < v->stream_count; i++) {
      if (ends[i] > so_start - v->ach.all_string_data_length) return -1;
      v->stream_offsets[i] = (uint32_t)ends[i];
    }
  }

  /* walk columns to bind bitmaps and stream slices */
  const uint32_t bitmap_size = (v->rows + 7) / 8;
  size_t pos = hp;
  uint32_t si = 0, str_off = 0;
  for (uint32_t c = 0; c < v->ncols; c++) {
    obx_cs_col_view *cv = &v->col[c];
    if (cv->h.attrs & OBX_CS_CA_HAS_NULL_BITMAP) {
      if (pos + bitmap_size > len) return -1;
      cv->null_bitmap = buf + pos;
      pos += bitmap_size;
    }
    if (cv->h.type == OBX_CS_COL_INTEGER) {
      if (si >= v->stream_count || v->stream_offsets[si] < pos) return -1;
      cv->int_stream = buf + pos;
      cv->int_stream_len = v->stream_offsets[si] - pos;
      pos = v->stream_offsets[si++];
    } else if (cv->h.type == OBX_CS_COL_STRING) {
      if (si >= v->stream_count || v->stream_offsets[si] < pos) return -1;
      int hn = obx_cs_str_meta_dec(buf + pos, v->stream_offsets[si] - pos,
                                   &cv->sm);
      if (hn < 0 || pos + (size_t)hn != v->stream_offsets[si]) return -1;
      pos = v->stream_offsets[si++];
      cv->str_data_off = str_off;
      str_off += cv->sm.uncompressed_len;
      if (str_off > v->ach.all_string_data_length) return -1;
      if (!(cv->sm.attr & OBX_CS_STR_FIXED_LEN)) {
        if (si >= v->stream_count || v->stream_offsets[si] < pos)
          return -1;
        cv->off_stream = buf + pos;
        cv->off_stream_len = v->stream_offsets[si] - pos;
        pos = v->stream_offsets[si++];
      }
    } else if (cv->h.type == OBX_CS_COL_INT_DICT ||
               cv->h.type == OBX_CS_COL_STR_DICT) {
      if (pos + sizeof(obx_cs_dict_meta) > len) return -1;
      memcpy(&cv->dm, buf + pos, sizeof(obx_cs_dict_meta));
      pos += sizeof(obx_cs_dict_meta);
      if (cv->dm.distinct_val_cnt == 0) continue; /* all-null column */
      if (cv->h.type == OBX_CS_COL_INT_DICT) {
        if (si >= v->stream_count || v->stream_offsets[si] < pos)
          return -1;
        cv->int_stream = buf + pos;
        cv->int_stream_len = v->stream_offsets[si] - pos;
        pos = v->stream_offsets[si++];
      } else {
        if (si >= v->stream_count || v->stream_offsets[si] < pos)
          return -1;
        int hn = obx_cs_str_meta_dec(buf + pos,
                                     v->stream_offsets[si] - pos,
                                     &cv->sm);
        if (hn < 0 || pos + (size_t)hn != v->stream_offsets[si])
          return -1;
        pos = v->stream_offsets[si++];
        cv->str_data_off = str_off;
        str_off += cv->sm.uncompressed_len;
        if (str_off > v->ach.all_string_data_length) return -1;
        if (!(cv->sm.attr & OBX_CS_STR_FIXED_LEN)) {
          if (si >= v->stream_count || v->stream_offsets[si] < pos)
            return -1;
          cv->off_stream = buf + pos;
          cv->off_stream_len = v->stream_offsets[si] - pos;
          pos = v->stream_offsets[si++];
        }
      }
      if (si >= v->stream_count || v->stream_offsets[si] < pos) return -1;
      cv->ref_stream = buf + pos;
      cv->ref_stream_len = v->stream_offsets[si] - pos;
      pos = v->stream_offsets[si++];
    } else {
      return -1;
    }
  }
  return 0;
}

static inline int blk_null(const uint8_t *bm, uint32_t r) {
  return bm && ((bm[r / 8] >> (7 - r % 8)) & 1); /* MSB-first in block */
}

/* decode a dict column's ref stream into one ref per row
 * (do_store_dict_ref_: plain row refs, or the const layout
 * [exception_cnt][const_ref][row ids][refs]) */
static int dict_refs(const obx_cs_col_view *cv, uint32_t rows,
                     uint32_t *refs) {
  uint64_t *ra = (uint64_t *)malloc((size_t)cv->dm.ref_row_cnt * 8);
  if (!ra) return -1;
  if (offset_stream_dec(cv->ref_stream, cv->ref_stream_len,
                        cv->dm.ref_row_cnt, ra) < 0) {
    free(ra);
    return -1;
  }
  if (cv->dm.attrs & OBX_CS_DICT_CONST_REF) {
    uint64_t ec = ra[0];
    if (cv->dm.ref_row_cnt != 2 + 2 * ec) { free(ra); return -1; }
    for (uint32_t r = 0; r < rows; r++) refs[r] = (uint32_t)ra[1];
    for (uint64_t i = 0; i < ec; i++) {
      uint64_t row = ra[2 + i];
      if (row >= rows) { free(ra); return -1; }
      refs[row] = (uint32_t)ra[2 + ec + i];
    }
  } else {
    if (cv->dm.ref_row_cnt != rows) { free(ra); return -1; }
    for (uint32_t r = 0; r < rows; r++) refs[r] = (uint32_t)ra[r];
  }
  free(ra);
  return 0;
}

int obx_cs_block_get_int(const obx_cs_block_view *v, uint32_t c,
                         int64_t *out, uint8_t *nulls_out) {
  if (c >= v->ncols) return -1;
  if (v->col[c].h.type == OBX_CS_COL_INT_DICT) {
    const obx_cs_col_view *cv = &v->col[c];
    uint32_t distinct = cv->dm.distinct_val_cnt;
    if (nulls_out) memset(nulls_out, 0, (v->rows + 7) / 8);
    if (distinct == 0) { /* all null */
      for (uint32_t r = 0; r < v->rows; r++) {
        out[r] = 0;
        if (nulls_out) nulls_out[r >> 3] |= (uint8_t)(1 << (r & 7));
      }
      return 0;
    }
    int64_t *dv = (int64_t *)malloc((size_t)distinct * 8);
    uint32_t *refs = (uint32_t *)malloc((size_t)v->rows * 4);
    int rc = -1;
    if (dv && refs &&
        obx_cs_int_stream_dec(cv->int_stream, cv->int_stream_len,
                              distinct, dv, NULL) >= 0 &&
        dict_refs(cv, v->rows, refs) == 0) {
      rc = 0;
      for (uint32_t r = 0; r < v->rows; r++) {
        uint32_t ref = refs[r];
        if (ref == distinct && (cv->dm.attrs & OBX_CS_DICT_HAS_NULL)) {
          out[r] = 0;
          if (nulls_out) nulls_out[r >> 3] |= (uint8_t)(1 << (r & 7));
        } else if (ref < distinct) {
          out[r] = dv[ref];
        } else {
          rc = -1;
          break;
        }
      }
    }
    free(dv);
    free(refs);
    return rc;
  }
  if (v->col[c].h.type != OBX_CS_COL_INTEGER) return -1;
  const obx_cs_col_view *cv = &v->col[c];
  obx_cs_int_meta m;
  if (obx_cs_int_stream_dec(cv->int_stream, cv->int_stream_len, v->rows,
                            out, &m) < 0)
    return -1;
  if (nulls_out) memset(nulls_out, 0, (v->rows + 7) / 8);
  if (m.attr & OBX_CS_REPLACE_NULL) {
    for (uint32_t r = 0; r < v->rows; r++)
      if (out[r] == (int64_t)m.null_replaced) {
        out[r] = 0;
        if (nulls_out) nulls_out[r >> 3] |= (uint8_t)(1 << (r & 7));
      }
  } else if (cv->null_bitmap) {
    for (uint32_t r = 0; r < v->rows; r++)
      if (blk_null(cv->null_bitmap, r)) {
        out[r] = 0;
        if (nulls_out) nulls_out[r >> 3] |= (uint8_t)(1 << (r & 7));
      }
  }
  return 0;
}

int64_t obx_cs_block_get_str(const obx_cs_block_view *v, uint32_t c,
                             uint8_t *bytes_out, size_t bytes_cap,
                             uint32_t *lens_out, uint8_t *nulls_out) {
  if (c >= v->ncols) return -1;
  if (v->col[c].h.type == OBX_CS_COL_STR_DICT) {
    const obx_cs_col_view *cv = &v->col[c];
    uint32_t distinct = cv->dm.distinct_val_cnt;
    if (nulls_out) memset(nulls_out, 0, (v->rows + 7) / 8);
    if (distinct == 0) {
      for (uint32_t r = 0; r < v->rows; r++) {
        lens_out[r] = 0;
        if (nulls_out) nulls_out[r >> 3] |= (uint8_t)(1 << (r & 7));
      }
      return 0;
    }
    /* dict value starts/lens from the pooled bytes */
    uint64_t *ends = (uint64_t *)malloc((size_t)distinct * 8);
    uint32_t *refs = (uint32_t *)malloc((size_t)v->rows * 4);
    int64_t rc = -1;
    if (!ends || !refs) goto done;
    if (cv->sm.attr & OBX_CS_STR_FIXED_LEN) {
      for (uint32_t i = 0; i < distinct; i++)
        ends[i] = (uint64_t)(i + 1) * cv->sm.fixed_str_len;
    } else if (offset_stream_dec(cv->off_stream, cv->off_stream_len,
                                 distinct, ends) < 0) {
      goto done;
    }
    if (ends[distinct - 1] != cv->sm.uncompressed_len) goto done;
    if (dict_refs(cv, v->rows, refs) != 0) goto done;
    {
      const uint8_t *dict_bytes = v->all_string + cv->str_data_off;
      size_t opos = 0;
      for (uint32_t r = 0; r < v->rows; r++) {
        uint32_t ref = refs[r];
        if (ref == distinct && (cv->dm.attrs & OBX_CS_DICT_HAS_NULL)) {
          lens_out[r] = 0;
          if (nulls_out) nulls_out[r >> 3] |= (uint8_t)(1 << (r & 7));
        } else if (ref < distinct) {
          uint64_t s = ref ? ends[ref - 1] : 0;
          uint32_t l = (uint32_t)(ends[ref] - s);
          if (opos + l > bytes_cap) goto done;
          memcpy(bytes_out + opos, dict_bytes + s, l);
          opos += l;
          lens_out[r] = l;
        } else {
          goto done;
        }
      }
      rc = (int64_t)opos;
    }
  done:
    free(ends);
    free(refs);
    return rc;
  }
  if (v->col[c].h.type != OBX_CS_COL_STRING) return -1;
  const obx_cs_col_view *cv = &v->col[c];
  const uint8_t *src = v->all_string + cv->str_data_off;
  uint32_t total = cv->sm.uncompressed_len;
  if ((size_t)total > bytes_cap) return -1;
  if (nulls_out) memset(nulls_out, 0, (v->rows + 7) / 8);
  if (cv->sm.attr & OBX_CS_STR_FIXED_LEN) {
    memcpy(bytes_out, src, total);
    for (uint32_t r = 0; r < v->rows; r++) {
      lens_out[r] = cv->sm.fixed_str_len;
      if (nulls_out && blk_null(cv->null_bitmap, r))
        nulls_out[r >> 3] |= (uint8_t)(1 << (r & 7));
    }
  } else {
    uint64_t *ends = (uint64_t *)malloc((size_t)v->rows * 8);
    if (!ends) return -1;
    if (offset_stream_dec(cv->off_stream, cv->off_stream_len, v->rows,
                          ends) < 0) {
      free(ends);
      return -1;
    }
    memcpy(bytes_out, src, total);
    uint64_t prev = 0;
    for (uint32_t r = 0; r < v->rows; r++) {
      if (ends[r] < prev || ends[r] > total) { free(ends); return -1; }
      lens_out[r] = (uint32_t)(ends[r] - prev);
      prev = ends[r];
      if (nulls_out) {
        int isnull = cv->null_bitmap
                         ? blk_null(cv->null_bitmap, r)
                         : ((cv->sm.attr & OBX_CS_STR_ZERO_LEN_NULL) &&
                            lens_out[r] == 0);
        if (isnull) nulls_out[r >> 3] |= (uint8_t)(1 << (r & 7));
      }
    }
    free(ends);
  }
  return (int64_t)total;
}
