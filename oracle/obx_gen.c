/*
 * obx_gen.c — seeded synthetic TPC-H lineitem generator (microblock writer).
 *
 * Produces the BASELINE.json config workloads as PAX microblocks in the
 * reference byte format (via obx_encode_block). Value distributions follow
 * dbgen's lineitem shapes (uniform quantity 1..50, discount 0.00..0.10,
 * tax 0.00..0.08, shipdate = orderdate + U[1,121], returnflag from
 * receiptdate vs 1995-06-17, linestatus from shipdate vs 1995-06-17,
 * 7 lineitems per order sharing an orderdate); exact dbgen streams are not
 * replicated (synthetic data per the benchmark contract — stated in
 * DESIGN.md). Deterministic in (seed, global_row_id): any shard of the row
 * range generates identical bytes on any rank.
 *
 * Dates are ObDateType day numbers since 1970-01-01
 * (/root/reference/deps/oblib/src/common/object/ob_obj_type.h ObDateType).
 * Decimals are scaled int64 decimal-int datums
 * (/root/reference/deps/oblib/src/lib/ob_define.h:1934-38: precision<=18 ->
 * int64).
 */
#include "obx_format.h"
#include "../include/obx.h"

#include <pthread.h>
#include <stdatomic.h>
#include <stdlib.h>
#include <string.h>
#include <unistd.h>

/* days from civil date (Howard Hinnant's algorithm), 1970-01-01 = 0 */
static int64_t days_from_civil(int y, int m, int d) {
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 399) / 400;
  unsigned yoe = (unsigned)(y - era * 400);
  unsigned doy = (unsigned)((153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1);
  unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return era * 146097 + (int64_t)doe - 719468;
}

static inline uint64_t mix64(uint64_t x) { /* splitmix64 finalizer */
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}
static inline uint64_t h2(uint64_t seed, uint64_t a, uint64_t b) {
  return mix64(seed ^ mix64(a ^ mix64(b)));
}

/* one lineitem row (scaled integer units) */
typedef struct li_row {
  int32_t shipdate;       /* days since 1970 */
  int64_t quantity2;      /* scale 2 units (e.g. 2400 = 24.00) */
  int64_t extprice2;      /* scale 2 */
  int64_t discount2;      /* scale 2: 0..10 */
  int64_t tax2;           /* scale 2: 0..8 */
  uint8_t returnflag, linestatus;
} li_row;

static void gen_row(uint64_t seed, uint64_t row, li_row *o) {
  const int64_t START = days_from_civil(1992, 1, 1);
  const int64_t END = days_from_civil(1998, 12, 1);
  const int64_t CURRENT = days_from_civil(1995, 6, 17);
  uint64_t order = row / 7;
  int64_t orderdate = START + (int64_t)(h2(seed, order, 1) % (uint64_t)(END - START - 151));
  uint64_t r1 = h2(seed, row, 2);
  uint64_t r2 = h2(seed, row, 3);
  uint64_t r3 = h2(seed, row, 4);
  uint64_t r4 = h2(seed, row, 5);
  int64_t ship = orderdate + 1 + (int64_t)(r1 % 121);
  int64_t receipt = ship + 1 + (int64_t)(r2 % 30);
  o->shipdate = (int32_t)ship;
  int64_t qty = 1 + (int64_t)(r3 % 50);
  o->quantity2 = qty * 100;
  int64_t price2 = 90000 + (int64_t)(r4 % 110001); /* 900.00 .. 2000.00 */
  o->extprice2 = qty * price2;
  o->discount2 = (int64_t)(h2(seed, row, 6) % 11);
  o->tax2 = (int64_t)(h2(seed, row, 7) % 9);
  o->returnflag = receipt <= CURRENT ? ((r1 >> 32) & 1 ? 'R' : 'A') : 'N';
  o->linestatus = ship > CURRENT ? 'O' : 'F';
}

/* ---- config column layouts ------------------------------ */
typedef struct gen_layout {
  uint16_t n_cols;
  obx_col_schema cols[8];
  uint8_t enc[8];
  int bytes_per_row_est;  /* to size rows-per-block */
} gen_layout;

static int layout_of(int config, gen_layout *L) {
  memset(L, 0, sizeof(*L));
  switch (config) {
    case 2: /* 1 col decimal-int64 quantity (1..50), RAW fixed 8 B */
      L->n_cols = 1;
      L->cols[0] = (obx_col_schema){ OBX_T_DECIMAL_INT, 0, 15, 8 };
      L->enc[0] = OBX_ENC_RAW;
      L->bytes_per_row_est = 8;
      return 0;
    case 3: /* shipdate INT_DIFF, quantity BIGINT RAW(bitpack), linestatus
               RLE, discount dict */
      L->n_cols = 4;
      L->cols[0] = (obx_col_schema){ OBX_T_DATE, 0, 0, 4 };
      L->enc[0] = OBX_ENC_INTEGER_BASE_DIFF;
      L->cols[1] = (obx_col_schema){ OBX_T_INT, 0, 19, 8 };
      L->enc[1] = OBX_ENC_RAW;
      L->cols[2] = (obx_col_schema){ OBX_T_CHAR, 0, 0, 1 };
      L->enc[2] = OBX_ENC_RLE;
      L->cols[3] = (obx_col_schema){ OBX_T_DECIMAL_INT, 2, 15, 8 };
      L->enc[3] = OBX_ENC_DICT;
      L->bytes_per_row_est = 4; /* ~12b + 6b + ~1B + 4b */
      return 0;
    case 4: /* Q1: quantity dict, extendedprice RAW 8B, discount dict,
               tax dict, returnflag dict, linestatus dict, shipdate diff */
      L->n_cols = 7;
      L->cols[0] = (obx_col_schema){ OBX_T_DECIMAL_INT, 2, 15, 8 };
      L->enc[0] = OBX_ENC_DICT;          /* l_quantity, 50 values */
      L->cols[1] = (obx_col_schema){ OBX_T_DECIMAL_INT, 2, 15, 8 };
      L->enc[1] = OBX_ENC_RAW;           /* l_extendedprice fixed 8 B */
      L->cols[2] = (obx_col_schema){ OBX_T_DECIMAL_INT, 2, 15, 8 };
      L->enc[2] = OBX_ENC_DICT;          /* l_discount, 11 values */
      L->cols[3] = (obx_col_schema){ OBX_T_DECIMAL_INT, 2, 15, 8 };
      L->enc[3] = OBX_ENC_DICT;          /* l_tax, 9 values */
      L->cols[4] = (obx_col_schema){ OBX_T_CHAR, 0, 0, 1 };
      L->enc[4] = OBX_ENC_DICT;          /* l_returnflag, 3 values */
      L->cols[5] = (obx_col_schema){ OBX_T_CHAR, 0, 0, 1 };
      L->enc[5] = OBX_ENC_DICT;          /* l_linestatus, 2 values */
      L->cols[6] = (obx_col_schema){ OBX_T_DATE, 0, 0, 4 };
      L->enc[6] = OBX_ENC_INTEGER_BASE_DIFF; /* l_shipdate ~12 bits */
      L->bytes_per_row_est = 12;
      return 0;
    case 6: /* Q6: shipdate diff, discount dict, quantity dict,
               extendedprice RAW 8B */
      L->n_cols = 4;
      L->cols[0] = (obx_col_schema){ OBX_T_DATE, 0, 0, 4 };
      L->enc[0] = OBX_ENC_INTEGER_BASE_DIFF;
      L->cols[1] = (obx_col_schema){ OBX_T_DECIMAL_INT, 2, 15, 8 };
      L->enc[1] = OBX_ENC_DICT;
      L->cols[2] = (obx_col_schema){ OBX_T_DECIMAL_INT, 2, 15, 8 };
      L->enc[2] = OBX_ENC_DICT;
      L->cols[3] = (obx_col_schema){ OBX_T_DECIMAL_INT, 2, 15, 8 };
      L->enc[3] = OBX_ENC_RAW;
      L->bytes_per_row_est = 11;
      return 0;
    default:
      return OBX_INVALID_ARGUMENT;
  }
}

/* fill column arrays for rows [row0, row0+n) */
static void fill_cols(int config, uint64_t seed, uint64_t row0, uint32_t n,
                      uint8_t **col_data, const gen_layout *L) {
  for (uint32_t i = 0; i < n; i++) {
    li_row lr;
    gen_row(seed, row0 + i, &lr);
    switch (config) {
      case 2: {
        int64_t q = lr.quantity2 / 100; /* scale 0: 1..50 */
        memcpy(col_data[0] + (size_t)i * 8, &q, 8);
        break;
      }
      case 3: {
        memcpy(col_data[0] + (size_t)i * 4, &lr.shipdate, 4);
        int64_t q = lr.quantity2 / 100;
        memcpy(col_data[1] + (size_t)i * 8, &q, 8);
        col_data[2][i] = lr.linestatus;
        memcpy(col_data[3] + (size_t)i * 8, &lr.discount2, 8);
        break;
      }
      case 4: {
        memcpy(col_data[0] + (size_t)i * 8, &lr.quantity2, 8);
        memcpy(col_data[1] + (size_t)i * 8, &lr.extprice2, 8);
        memcpy(col_data[2] + (size_t)i * 8, &lr.discount2, 8);
        memcpy(col_data[3] + (size_t)i * 8, &lr.tax2, 8);
        col_data[4][i] = lr.returnflag;
        col_data[5][i] = lr.linestatus;
        memcpy(col_data[6] + (size_t)i * 4, &lr.shipdate, 4);
        break;
      }
      case 6: {
        memcpy(col_data[0] + (size_t)i * 4, &lr.shipdate, 4);
        memcpy(col_data[1] + (size_t)i * 8, &lr.discount2, 8);
        memcpy(col_data[2] + (size_t)i * 8, &lr.quantity2, 8);
        memcpy(col_data[3] + (size_t)i * 8, &lr.extprice2, 8);
        break;
      }
    }
  }
}

typedef struct blk_loc {
  int64_t size;
  int32_t worker;
  int64_t arena_off;
} blk_loc;

typedef struct gen_job {
  int config;
  uint64_t seed, row_count, row_id_base;
  uint32_t rows_per_block;
  uint64_t n_blocks;
  int64_t slot_bytes;      /* per-block worst-case bound */
  blk_loc *locs;           /* per-block location in its worker's arena */
  uint8_t **arenas;        /* per-worker arena (realloc-grown) */
  int64_t *arena_len;
  const gen_layout *L;
  _Atomic uint64_t *next;
  _Atomic int *err;
  int worker_id;
} gen_job;

static void *gen_worker(void *arg) {
  gen_job *J = (gen_job *)arg;
  const gen_layout *L = J->L;
  const int w = J->worker_id;
  uint8_t *cbuf[8];
  for (int c = 0; c < L->n_cols; c++)
    cbuf[c] = (uint8_t *)malloc((size_t)J->rows_per_block * L->cols[c].len);
  int64_t cap = 0, len = 0;
  uint8_t *arena = NULL;
  for (;;) {
    uint64_t b = atomic_fetch_add(J->next, 1);
    if (b >= J->n_blocks) break;
    uint64_t row0 = J->row_id_base + b * J->rows_per_block;
    uint32_t n = (uint32_t)((b == J->n_blocks - 1)
                                ? J->row_count - b * J->rows_per_block
                                : J->rows_per_block);
    fill_cols(J->config, J->seed, row0, n, cbuf, L);
    if (len + J->slot_bytes > cap) {
      cap = cap ? cap * 2 : (J->slot_bytes * 1024);
      arena = (uint8_t *)realloc(arena, (size_t)cap);
      if (!arena) { atomic_store(J->err, OBX_INTERNAL_ERROR); break; }
    }
    int64_t sz = obx_encode_block(L->cols, L->n_cols,
                                  (const uint8_t *const *)cbuf, NULL, n,
                                  L->enc, arena + len, J->slot_bytes);
    if (sz < 0) { atomic_store(J->err, (int)sz); break; }
    J->locs[b].size = sz;
    J->locs[b].worker = w;
    J->locs[b].arena_off = len;
    len += sz;
  }
  J->arenas[w] = arena;
  J->arena_len[w] = len;
  for (int c = 0; c < L->n_cols; c++) free(cbuf[c]);
  return NULL;
}

int64_t obx_gen_lineitem(int config, uint64_t row_count, uint64_t seed,
                         uint32_t target_block_bytes, uint64_t row_id_base,
                         uint8_t **out_data, uint64_t **out_offsets,
                         obx_col_schema *out_cols, uint16_t *out_n_cols) {
  gen_layout L;
  if (layout_of(config, &L)) return OBX_INVALID_ARGUMENT;
  if (!row_count || !out_data || !out_offsets) return OBX_INVALID_ARGUMENT;
  if (!target_block_bytes) target_block_bytes = 16384; /* 16 KB default
      (OB_DEFAULT_SSTABLE_BLOCK_SIZE, ob_define.h:2004) */
  uint32_t rpb = (uint32_t)((target_block_bytes - 512) / L.bytes_per_row_est);
  if (rpb < 16) rpb = 16;
  if (rpb > 65535) rpb = 65535;
  uint64_t n_blocks = (row_count + rpb - 1) / rpb;
  /* per-block worst-case bound (encoded blocks are ~target size; forced
     encodings keep this tight) */
  int64_t slot = 2 * (int64_t)target_block_bytes + 4096;

  int nthreads = (int)sysconf(_SC_NPROCESSORS_ONLN);
  const char *envt = getenv("OBX_GEN_THREADS");
  if (envt && atoi(envt) > 0) nthreads = atoi(envt);
  if (nthreads < 1) nthreads = 1;
  if ((uint64_t)nthreads > n_blocks) nthreads = (int)n_blocks;
  if (nthreads > 256) nthreads = 256;

  blk_loc *locs = (blk_loc *)calloc(n_blocks, sizeof(blk_loc));
  uint8_t **arenas = (uint8_t **)calloc((size_t)nthreads, sizeof(uint8_t *));
  int64_t *arena_len = (int64_t *)calloc((size_t)nthreads, sizeof(int64_t));
  gen_job *jobs = (gen_job *)calloc((size_t)nthreads, sizeof(gen_job));
  if (!locs || !arenas || !arena_len || !jobs) {
    free(locs); free(arenas); free(arena_len); free(jobs);
    return OBX_INTERNAL_ERROR;
  }
  _Atomic uint64_t next = 0;
  _Atomic int err = 0;
  for (int t = 0; t < nthreads; t++) {
    jobs[t].config = config; jobs[t].seed = seed;
    jobs[t].row_count = row_count; jobs[t].row_id_base = row_id_base;
    jobs[t].rows_per_block = rpb; jobs[t].n_blocks = n_blocks;
    jobs[t].slot_bytes = slot; jobs[t].locs = locs;
    jobs[t].arenas = arenas; jobs[t].arena_len = arena_len;
    jobs[t].L = &L; jobs[t].next = &next; jobs[t].err = &err;
    jobs[t].worker_id = t;
  }
  pthread_t tids[256];
  for (int t = 1; t < nthreads; t++)
    pthread_create(&tids[t], NULL, gen_worker, &jobs[t]);
  gen_worker(&jobs[0]);
  for (int t = 1; t < nthreads; t++) pthread_join(tids[t], NULL);
  if (atomic_load(&err)) {
    for (int t = 0; t < nthreads; t++) free(arenas[t]);
    free(locs); free(arenas); free(arena_len); free(jobs);
    return atomic_load(&err);
  }

  uint64_t *offs = (uint64_t *)malloc(sizeof(uint64_t) * (n_blocks + 1));
  int64_t total = 0;
  for (uint64_t b = 0; b < n_blocks; b++) {
    offs[b] = (uint64_t)total;
    /* 16-B align each block in the container (the GPU engine stages blocks
       to LDS with dwordx4 copies; macro-block writers align similarly) */
    total = (total + locs[b].size + 15) & ~(int64_t)15;
  }
  offs[n_blocks] = (uint64_t)total;
  uint8_t *data = (uint8_t *)malloc((size_t)total + 16);
  if (!offs || !data) {
    for (int t = 0; t < nthreads; t++) free(arenas[t]);
    free(locs); free(arenas); free(arena_len); free(jobs);
    free(offs); free(data);
    return OBX_INTERNAL_ERROR;
  }
  for (uint64_t b = 0; b < n_blocks; b++) {
    memcpy(data + offs[b], arenas[locs[b].worker] + locs[b].arena_off,
           (size_t)locs[b].size);
    int64_t pad = (int64_t)offs[b + 1] - (int64_t)offs[b] - locs[b].size;
    if (pad > 0) memset(data + offs[b] + locs[b].size, 0, (size_t)pad);
  }
  memset(data + total, 0, 16); /* slack for 9-byte bitstream reads */
  for (int t = 0; t < nthreads; t++) free(arenas[t]);
  free(locs); free(arenas); free(arena_len); free(jobs);
  *out_data = data;
  *out_offsets = offs;
  if (out_cols) memcpy(out_cols, L.cols, sizeof(obx_col_schema) * L.n_cols);
  if (out_n_cols) *out_n_cols = L.n_cols;
  return (int64_t)n_blocks;
}

/* date helper exported for tests/bench (e.g. '1998-09-02' -> day number) */
int64_t obx_date_days(int y, int m, int d) { return days_from_civil(y, m, d); }
