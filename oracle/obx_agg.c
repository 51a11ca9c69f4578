/*
 * obx_agg.c — CPU ORACLE fused scan->filter->aggregate (and the host-cores
 * baseline timed by bench.py's cpu_baseline leg).
 *
 * Restates the reference pipeline (all paths under /root/reference/):
 *   driver loop     storage/access/ob_vector_store.cpp:329-389 +
 *                   ob_sstable_row_scanner.cpp:557-598 (per-microblock
 *                   filter -> project -> consume)
 *   group-by        sql/engine/aggregate/ob_hash_groupby_vec_op.cpp:1400-1515
 *                   (find-or-create group row per input row; our table is a
 *                   flat array probe, the reference's is
 *                   ObExtendHashTableVec, ob_exec_hash_struct_vec.h:1718 —
 *                   result set is identical, order normalized by key sort)
 *   aggregation     share/aggregate/processor.h:47, sum.h:49-220 (COUNT int64;
 *                   SUM decimal accumulates in wide ints; we carry all sums in
 *                   256-bit two's-complement limbs, exact for every config)
 *   decimal exprs   sql/engine/expr/ob_expr_mul.cpp semantics:
 *                   disc_price = extprice * (1 - discount) and
 *                   charge = disc_price * (1 + tax) in scaled integer units
 *                   (scales 2 -> 4 -> 6), exact integer arithmetic
 *   output order:   groups sorted by raw key bytes (Q1's ORDER BY
 *                   l_returnflag, l_linestatus on char keys == memcmp order)
 *
 * Multi-threading mirrors PX block-granule splitting
 * (sql/engine/px/ob_granule_util.h:307): atomic microblock work-stealing,
 * per-thread partial tables, final merge — the same shape the multi-GPU
 * shard uses.
 */
#include "obx_format.h"
#include "../include/obx.h"

#include <pthread.h>
#include <stdatomic.h>
#include <stdlib.h>
#include <string.h>
#include <unistd.h>

/* from obx_codec.c */
int obx__col_dec_init(void *d, const void *h, const void *ch,
                      const uint8_t *meta_region);
int obx__col_dec_row(const void *d, const void *h, const obx_col_schema *cs,
                     uint32_t r, int64_t *out, int *is_null);
int obx__black_eval(const obx_filter_leaf *lf, const int64_t *vals,
                    const int *nulls);
int obx__bprog_valid(const obx_filter_leaf *lf);
int obx__leaf_match(const obx_filter_leaf *lf, int64_t v, int is_null, int sc,
                    int len);
int obx__combine_leaves(const obx_filter_desc *f, const int *leaf_res);
int obx__prog_valid(const obx_filter_desc *f);
size_t obx__col_dec_size(void);

/* ---- 256-bit two's-complement accumulation ------------------------------ */
typedef struct { uint64_t l[4]; } i256;

static inline void i256_add(i256 *a, const i256 *b) {
  unsigned __int128 c = 0;
  for (int i = 0; i < 4; i++) {
    c += (unsigned __int128)a->l[i] + b->l[i];
    a->l[i] = (uint64_t)c;
    c >>= 64;
  }
}
static inline void i256_from_i64(i256 *a, int64_t v) {
  a->l[0] = (uint64_t)v;
  uint64_t s = v < 0 ? ~(uint64_t)0 : 0;
  a->l[1] = a->l[2] = a->l[3] = s;
}
static inline void i256_from_i128(i256 *a, __int128 v) {
  a->l[0] = (uint64_t)v;
  a->l[1] = (uint64_t)((unsigned __int128)v >> 64);
  uint64_t s = v < 0 ? ~(uint64_t)0 : 0;
  a->l[2] = a->l[3] = s;
}
static inline void i256_add_i64(i256 *a, int64_t v) {
  i256 t; i256_from_i64(&t, v); i256_add(a, &t);
}
static inline void i256_add_i128(i256 *a, __int128 v) {
  i256 t; i256_from_i128(&t, v); i256_add(a, &t);
}
/* (int128 p) * (int64 m) -> int256, exact */
static inline void i256_add_mul_i128_i64(i256 *a, __int128 p, int64_t m) {
  int neg = 0;
  unsigned __int128 up;
  uint64_t um;
  if (p < 0) { up = (unsigned __int128)(-p); neg ^= 1; }
  else up = (unsigned __int128)p;
  if (m < 0) { um = (uint64_t)(-m); neg ^= 1; }
  else um = (uint64_t)m;
  uint64_t p0 = (uint64_t)up, p1 = (uint64_t)(up >> 64);
  unsigned __int128 m0 = (unsigned __int128)p0 * um;
  unsigned __int128 m1 = (unsigned __int128)p1 * um;
  i256 t;
  t.l[0] = (uint64_t)m0;
  unsigned __int128 mid = (m0 >> 64) + (unsigned __int128)(uint64_t)m1;
  t.l[1] = (uint64_t)mid;
  t.l[2] = (uint64_t)((mid >> 64) + (uint64_t)(m1 >> 64));
  t.l[3] = 0;
  if (neg) { /* two's complement negate */
    unsigned __int128 c = 1;
    for (int i = 0; i < 4; i++) {
      c += (unsigned __int128)(~t.l[i]);
      t.l[i] = (uint64_t)c;
      c >>= 64;
    }
  }
  i256_add(a, &t);
}

/* ---- scale constants ---------------------------------------------------- */
static const int64_t POW10[19] = {
  1ll,10ll,100ll,1000ll,10000ll,100000ll,1000000ll,10000000ll,100000000ll,
  1000000000ll,10000000000ll,100000000000ll,1000000000000ll,10000000000000ll,
  100000000000000ll,1000000000000000ll,10000000000000000ll,
  100000000000000000ll,1000000000000000000ll };

/* ---- per-thread aggregation state --------------------------------------- */
typedef struct grp {
  uint8_t key[OBX_MAX_KEY_BYTES];
  uint8_t key_len;
  int used;
  uint64_t row_count;
  i256 cell[8];
  int has_val[8]; /* for MIN/MAX init */
} grp;

#define OBX_CPU_MAX_GROUPS 65536 /* growth cap (the reference's table
                                     grows unboundedly; this bounds test
                                     memory) */
typedef struct work_ctx {
  const obx_blockset *bs;
  const obx_filter_desc *filter;
  const obx_agg_desc *agg;
  _Atomic uint64_t *next_block;
  grp *groups;          /* grown on demand (ob_exec_hash_struct_vec.h:1718
                           grows unboundedly) */
  uint32_t n_groups, gcap;
  uint64_t rows_scanned, rows_passed;
  int rc;
} work_ctx;

/* key bytes = concatenated datums (zeros for NULL) + one null-flags byte
 * (bit per group col) appended at key[key_len] — NULL forms its own group
 * without colliding with value 0 (mirrors the GPU key layout). */
static grp *find_group(work_ctx *w, const uint8_t *key, uint8_t key_len) {
  for (uint32_t i = 0; i < w->n_groups; i++) {
    grp *g = &w->groups[i];
    if (g->key_len == key_len && memcmp(g->key, key, (size_t)key_len + 1) == 0)
      return g;
  }
  if (w->n_groups >= w->gcap) {
    uint32_t nc = w->gcap ? w->gcap * 2 : OBX_MAX_GROUPS;
    if (nc > OBX_CPU_MAX_GROUPS) return NULL;
    grp *ng = (grp *)realloc(w->groups, (size_t)nc * sizeof(grp));
    if (!ng) return NULL;
    w->groups = ng;
    w->gcap = nc;
  }
  grp *g = &w->groups[w->n_groups++];
  memset(g, 0, sizeof(*g));
  memcpy(g->key, key, (size_t)key_len + 1);
  g->key_len = key_len;
  g->used = 1;
  return g;
}

static void *worker(void *arg) {
  work_ctx *w = (work_ctx *)arg;
  const obx_blockset *bs = w->bs;
  const obx_agg_desc *agg = w->agg;
  const obx_filter_desc *filter = w->filter;
  size_t dsz = obx__col_dec_size();
  uint8_t *decbuf = (uint8_t *)malloc(dsz * 32);
  if (!decbuf) { w->rc = OBX_INTERNAL_ERROR; return NULL; }

  /* which columns do we need decoders for? */
  uint16_t need[32]; int n_need = 0;
  int dec_of_col[64];
  for (int i = 0; i < 64; i++) dec_of_col[i] = -1;
#define NEED(c) do { uint16_t cc = (c); \
  if (cc < 64 && dec_of_col[cc] < 0) { dec_of_col[cc] = n_need; need[n_need++] = cc; } } while (0)
  if (filter) for (int i = 0; i < filter->n_leaves; i++) {
    const obx_filter_leaf *lf0 = &filter->leaves[i];
    if (lf0->op == OBX_OP_BLACK) {
      for (int j = 0; j < lf0->n_bcols; j++) {
        if (lf0->bcols[j] >= bs->n_cols) {
          w->rc = OBX_INVALID_ARGUMENT;
          free(decbuf);
          return NULL;
        }
        NEED(lf0->bcols[j]);
      }
    } else {
      if (lf0->col >= bs->n_cols) {
        w->rc = OBX_INVALID_ARGUMENT;
        free(decbuf);
        return NULL;
      }
      NEED(lf0->col);
    }
  }
  if (agg) {
    for (int i = 0; i < agg->n_group_cols; i++) {
      if (agg->group_cols[i] >= bs->n_cols) {
        w->rc = OBX_INVALID_ARGUMENT;
        free(decbuf);
        return NULL;
      }
      NEED(agg->group_cols[i]);
    }
    for (int i = 0; i < agg->n_aggs; i++) {
      const obx_agg_expr *e = &agg->aggs[i];
      if (e->col_a != UINT16_MAX && e->col_a >= bs->n_cols) {
        w->rc = OBX_INVALID_ARGUMENT;
        free(decbuf);
        return NULL;
      }
      if (e->col_a != UINT16_MAX) NEED(e->col_a);
      if (e->kind == OBX_AGG_SUM_PROD2 || e->kind == OBX_AGG_SUM_PROD3 ||
          e->kind == OBX_AGG_SUM_MUL) {
        if (e->col_b >= bs->n_cols) {
          w->rc = OBX_INVALID_ARGUMENT;
          free(decbuf);
          return NULL;
        }
        NEED(e->col_b);
      }
      if (e->kind == OBX_AGG_SUM_PROD3) {
        if (e->col_c >= bs->n_cols) {
          w->rc = OBX_INVALID_ARGUMENT;
          free(decbuf);
          return NULL;
        }
        NEED(e->col_c);
      }
    }
  }
#undef NEED

  for (;;) {
    uint64_t b = atomic_fetch_add(w->next_block, 1);
    if (b >= bs->n_blocks) break;
    const uint8_t *block = bs->data + bs->block_offsets[b];
    const obx_micro_header *h = (const obx_micro_header *)block;
    /* read == verify (negligible: HW CRC-32C at ~20 GB/s/core) */
    int64_t blen = (int64_t)(bs->block_offsets[b + 1] - bs->block_offsets[b]);
    if (h->magic != OBX_MICRO_BLOCK_MAGIC ||
        (int64_t)h->header_size +
                (int64_t)bs->n_cols * (int64_t)sizeof(obx_col_header) >
            blen ||
        h->data_zlength > blen || h->data_zlength < OBX_MICRO_HEADER_SIZE) {
      w->rc = OBX_INVALID_ARGUMENT;
      free(decbuf);
      return NULL;
    }
    if ((int64_t)obx_crc32c(block + OBX_MICRO_HEADER_SIZE,
                            (int64_t)h->data_zlength -
                                OBX_MICRO_HEADER_SIZE) != h->data_checksum) {
      w->rc = OBX_PHYSIC_CHECKSUM_ERROR;
      free(decbuf);
      return NULL;
    }
    const uint8_t *meta_region = block + h->header_size +
        (int64_t)bs->n_cols * sizeof(obx_col_header);
    const obx_col_header *chp =
        (const obx_col_header *)(block + h->header_size);
    for (int i = 0; i < n_need; i++) {
      if (obx__col_dec_init(decbuf + dsz * i, h, &chp[need[i]], meta_region)) {
        w->rc = OBX_INTERNAL_ERROR; free(decbuf); return NULL;
      }
    }
    uint32_t rows = h->row_count;
    w->rows_scanned += rows;
    int64_t vals[32]; int nulls[32];
    for (uint32_t r = 0; r < rows; r++) {
      /* filter (conjunctive fast path or combine program) */
      int pass = 1;
      if (filter && filter->n_prog == 0) {
        for (int i = 0; i < filter->n_leaves && pass; i++) {
          const obx_filter_leaf *lf = &filter->leaves[i];
          if (lf->op == OBX_OP_BLACK) {
            int64_t bvv[OBX_BX_MAX_COLS]; int bnn[OBX_BX_MAX_COLS];
            for (int j = 0; j < lf->n_bcols; j++)
              obx__col_dec_row(decbuf + dsz * dec_of_col[lf->bcols[j]], h,
                               &bs->cols[lf->bcols[j]], r, &bvv[j],
                               &bnn[j]);
            pass = obx__black_eval(lf, bvv, bnn);
            continue;
          }
          int di = dec_of_col[lf->col];
          int64_t v; int isn;
          obx__col_dec_row(decbuf + dsz * di, h, &bs->cols[lf->col], r, &v,
                           &isn);
          pass = obx__leaf_match(lf, v, isn,
                                 obx_store_class(bs->cols[lf->col].obj_type),
                                 bs->cols[lf->col].len);
        }
      } else if (filter && filter->n_leaves > 0) {
        int leaf_res[8];
        for (int i = 0; i < filter->n_leaves; i++) {
          const obx_filter_leaf *lf = &filter->leaves[i];
          if (lf->op == OBX_OP_BLACK) {
            int64_t bvv[OBX_BX_MAX_COLS]; int bnn[OBX_BX_MAX_COLS];
            for (int j = 0; j < lf->n_bcols; j++)
              obx__col_dec_row(decbuf + dsz * dec_of_col[lf->bcols[j]], h,
                               &bs->cols[lf->bcols[j]], r, &bvv[j],
                               &bnn[j]);
            leaf_res[i] = obx__black_eval(lf, bvv, bnn);
            continue;
          }
          int di = dec_of_col[lf->col];
          int64_t v; int isn;
          obx__col_dec_row(decbuf + dsz * di, h, &bs->cols[lf->col], r, &v,
                           &isn);
          leaf_res[i] = obx__leaf_match(
              lf, v, isn, obx_store_class(bs->cols[lf->col].obj_type),
              bs->cols[lf->col].len);
        }
        pass = obx__combine_leaves(filter, leaf_res);
      }
      if (!pass) continue;
      w->rows_passed++;
      if (!agg) continue;
      /* decode remaining needed columns */
      for (int i = 0; i < n_need; i++) {
        obx__col_dec_row(decbuf + dsz * i, h, &bs->cols[need[i]], r, &vals[i],
                         &nulls[i]);
      }
      /* group key (+ null-flags byte at key[klen]) */
      uint8_t key[OBX_MAX_KEY_BYTES]; int klen = 0; uint8_t nflags = 0;
      for (int i = 0; i < agg->n_group_cols; i++) {
        uint16_t c = agg->group_cols[i];
        int di = dec_of_col[c];
        if (nulls[di]) {
          memset(key + klen, 0, bs->cols[c].len);
          nflags |= (uint8_t)(1u << i);
        } else {
          memcpy(key + klen, &vals[di], bs->cols[c].len);
        }
        klen += bs->cols[c].len;
      }
      key[klen] = nflags;
      grp *g = find_group(w, key, (uint8_t)klen);
      if (!g) { w->rc = OBX_BUF_NOT_ENOUGH; free(decbuf); return NULL; }
      g->row_count++;
      for (int i = 0; i < agg->n_aggs; i++) {
        const obx_agg_expr *e = &agg->aggs[i];
        switch (e->kind) {
          case OBX_AGG_COUNT: {
            if (e->col_a == UINT16_MAX || !nulls[dec_of_col[e->col_a]])
              i256_add_i64(&g->cell[i], 1);
            break;
          }
          case OBX_AGG_SUM: {
            int di = dec_of_col[e->col_a];
            if (!nulls[di]) i256_add_i64(&g->cell[i], vals[di]);
            break;
          }
          case OBX_AGG_MIN: case OBX_AGG_MAX: {
            int di = dec_of_col[e->col_a];
            if (!nulls[di]) {
              int64_t v = vals[di];
              int64_t cur = (int64_t)g->cell[i].l[0];
              if (!g->has_val[i] ||
                  (e->kind == OBX_AGG_MIN ? v < cur : v > cur)) {
                i256_from_i64(&g->cell[i], v);
                g->has_val[i] = 1;
              }
            }
            break;
          }
          case OBX_AGG_SUM_PROD2: {
            int da = dec_of_col[e->col_a], db = dec_of_col[e->col_b];
            if (!nulls[da] && !nulls[db]) {
              int64_t one = POW10[bs->cols[e->col_b].scale];
              __int128 p = (__int128)vals[da] * (one - vals[db]);
              i256_add_i128(&g->cell[i], p);
            }
            break;
          }
          case OBX_AGG_SUM_PROD3: {
            int da = dec_of_col[e->col_a], db = dec_of_col[e->col_b],
                dc = dec_of_col[e->col_c];
            if (!nulls[da] && !nulls[db] && !nulls[dc]) {
              int64_t one_b = POW10[bs->cols[e->col_b].scale];
              int64_t one_c = POW10[bs->cols[e->col_c].scale];
              __int128 p = (__int128)vals[da] * (one_b - vals[db]);
              i256_add_mul_i128_i64(&g->cell[i], p, one_c + vals[dc]);
            }
            break;
          }
          case OBX_AGG_SUM_MUL: {
            int da = dec_of_col[e->col_a], db = dec_of_col[e->col_b];
            if (!nulls[da] && !nulls[db]) {
              __int128 p = (__int128)vals[da] * vals[db];
              i256_add_i128(&g->cell[i], p);
            }
            break;
          }
          default: break;
        }
      }
    }
  }
  free(decbuf);
  return NULL;
}

static obx_group_row *g_last_rows = NULL;
static uint64_t g_last_n = 0;

int obx_cpu_scan_filter_agg(const obx_blockset *bs,
                            const obx_filter_desc *filter,
                            const obx_agg_desc *agg, int nthreads,
                            obx_agg_result *out) {
  if (!bs || !out) return OBX_INVALID_ARGUMENT;
  if (filter && !obx__prog_valid(filter)) return OBX_INVALID_ARGUMENT;
  if (filter)
    for (int i = 0; i < filter->n_leaves; i++)
      if (filter->leaves[i].op == OBX_OP_BLACK &&
          !obx__bprog_valid(&filter->leaves[i]))
        return OBX_INVALID_ARGUMENT;
  for (uint16_t c = 0; c < bs->n_cols; c++) {
    if (bs->cols[c].scale < 0 || bs->cols[c].scale > 18)
      return OBX_NOT_SUPPORTED; /* POW10 bound (decimal-int <= 18 digits) */
  }
  if (nthreads <= 0) nthreads = (int)sysconf(_SC_NPROCESSORS_ONLN);
  if (nthreads < 1) nthreads = 1;
  if (nthreads > 256) nthreads = 256;

  _Atomic uint64_t next = 0;
  work_ctx *ws = (work_ctx *)calloc((size_t)nthreads, sizeof(work_ctx));
  pthread_t *tids = (pthread_t *)malloc(sizeof(pthread_t) * (size_t)nthreads);
  if (!ws || !tids) { free(ws); free(tids); return OBX_INTERNAL_ERROR; }
  for (int t = 0; t < nthreads; t++) {
    ws[t].bs = bs; ws[t].filter = filter; ws[t].agg = agg;
    ws[t].next_block = &next;
  }
  for (int t = 1; t < nthreads; t++)
    pthread_create(&tids[t], NULL, worker, &ws[t]);
  worker(&ws[0]);
  for (int t = 1; t < nthreads; t++) pthread_join(tids[t], NULL);

  int rc = OBX_SUCCESS;
  memset(out, 0, sizeof(*out));
  /* merge partial tables (the reference's 2-phase group-by exchange merge;
     identical to the per-GPU RCCL partial merge in §8e) */
  work_ctx merged;
  memset(&merged, 0, sizeof(merged));
  for (int t = 0; t < nthreads; t++) {
    if (ws[t].rc) rc = ws[t].rc;
    out->rows_scanned += ws[t].rows_scanned;
    out->rows_passed += ws[t].rows_passed;
    for (uint32_t i = 0; i < ws[t].n_groups; i++) {
      grp *s = &ws[t].groups[i];
      grp *g = find_group(&merged, s->key, s->key_len);
      if (!g) { rc = OBX_BUF_NOT_ENOUGH; break; }
      g->row_count += s->row_count;
      int na = agg ? agg->n_aggs : 0;
      for (int a = 0; a < na; a++) {
        if (agg->aggs[a].kind == OBX_AGG_MIN || agg->aggs[a].kind == OBX_AGG_MAX) {
          if (s->has_val[a]) {
            int64_t v = (int64_t)s->cell[a].l[0];
            int64_t cur = (int64_t)g->cell[a].l[0];
            if (!g->has_val[a] ||
                (agg->aggs[a].kind == OBX_AGG_MIN ? v < cur : v > cur)) {
              g->cell[a] = s->cell[a];
              g->has_val[a] = 1;
            }
          }
        } else {
          i256_add(&g->cell[a], &s->cell[a]);
        }
      }
    }
  }
  /* sort groups by key bytes for deterministic output */
  for (uint32_t i = 1; i < merged.n_groups; i++) {
    grp tmp = merged.groups[i];
    uint32_t j = i;
    while (j > 0) {
      grp *p = &merged.groups[j - 1];
      int c = memcmp(p->key, tmp.key,
                     (size_t)(p->key_len < tmp.key_len ? p->key_len
                                                       : tmp.key_len) + 1);
      if (c > 0 || (c == 0 && p->key_len > tmp.key_len)) {
        merged.groups[j] = *p; j--;
      } else break;
    }
    merged.groups[j] = tmp;
  }
  /* cache ALL rows for paged access (obx_cpu_agg_fetch), surface the
     inline result when it fits */
  free(g_last_rows);
  g_last_rows = NULL;
  g_last_n = 0;
  if (merged.n_groups) {
    g_last_rows =
        (obx_group_row *)calloc(merged.n_groups, sizeof(obx_group_row));
    if (g_last_rows) {
      for (uint32_t i = 0; i < merged.n_groups; i++) {
        grp *s = &merged.groups[i];
        obx_group_row *o = &g_last_rows[i];
        memcpy(o->key, s->key, OBX_MAX_KEY_BYTES);
        o->key_len = s->key_len;
        o->row_count = s->row_count;
        for (int a = 0; a < 8; a++)
          memcpy(&o->cells[a], &s->cell[a], sizeof(obx_agg_cell));
      }
      g_last_n = merged.n_groups;
    }
  }
  out->n_groups = merged.n_groups;
  if (merged.n_groups <= OBX_MAX_GROUPS) {
    for (uint32_t i = 0; i < merged.n_groups; i++)
      out->groups[i] = g_last_rows[i];
  } else if (rc == OBX_SUCCESS) {
    rc = OBX_BUF_NOT_ENOUGH; /* rows stay paged behind obx_cpu_agg_fetch */
  }
  for (int t = 0; t < nthreads; t++) free(ws[t].groups);
  free(merged.groups);
  free(ws); free(tids);
  return rc;
}

/* paged access to the last scan's sorted group rows (single-threaded test
 * infrastructure; mirrors obx_gpu_agg_fetch) */
int obx_cpu_agg_fetch(uint32_t start, uint32_t count, obx_group_row *out,
                      uint32_t *n_out, uint64_t *n_total) {
  if (!out) return OBX_INVALID_ARGUMENT;
  if (n_total) *n_total = g_last_n;
  uint32_t n = 0;
  for (; n < count && start + n < g_last_n; n++)
    out[n] = g_last_rows[start + n];
  if (n_out) *n_out = n;
  return OBX_SUCCESS;
}
