/*
 * obx_kernels.hip — gfx950 (MI355X/CDNA4) kernels for the OLAP hot path:
 * microblock decode -> pushdown filter -> hash group-by aggregate.
 *
 * Design (MI355X-first, not a translation of the reference's AVX loops):
 *  - one 256-thread workgroup processes whole microblocks from a grid-stride
 *    loop; each block's encoded bytes are STAGED THROUGH LDS with coalesced
 *    dwordx4 copies (HBM sees exactly the encoded column pages, read once,
 *    fully coalesced), then all bit-granular decode reads hit LDS
 *    (~16 KB block << 160 KB LDS/CU). Blocks larger than
 *    OBX_LDS_STAGE_BYTES take a generic global-memory path.
 *  - white filters arrive pre-lowered per block (obx_dev.h): dict-domain
 *    64-bit ref masks or packed-domain unsigned ranges, so the inner loop is
 *    unpack + mask/range test + __ballot (cf. the reference's AVX512
 *    compares, ob_raw_decoder_simd.cpp, and dict-domain filters,
 *    ob_dict_decoder.cpp:810-886)
 *  - selection vectors are built with wavefront ballot + mbcnt prefix sums
 *    (the get_row_ids bitmap->row_ids compaction,
 *    ob_block_batched_row_store.cpp:107)
 *  - group-by: per-row group key -> wave-level cell clustering (ballot +
 *    64-wide shfl reductions of int128 partials) -> per-workgroup LDS table
 *    (int128 cells, carry-propagating u64 atomics) -> one global open-
 *    addressing table flush per workgroup (256-bit cells). Mirrors
 *    ObExtendHashTableVec + aggregate::Processor semantics
 *    (ob_exec_hash_struct_vec.h:1718, share/aggregate/sum.h) with exact
 *    integer arithmetic.
 *  - no MFMA anywhere: this path is memory-bound integer work.
 */
#include "obx_dev_common.h"

struct prog_spec { uint8_t n; uint8_t t[15]; };

/* cold path: evaluate a combine-program filter for one row window.
 * __noinline__ + by-value args keep its registers off the hot AND path
 * (called once per block window only when the plan has a program). */
__device__ __forceinline__ void filter_prog_phase(
    const blk_view bv, const dev_block &cur,
    const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bl, uint32_t n_leaves, prog_spec ps,
    uint32_t rows, uint32_t w0, uint64_t blk_bit, lds3_u64 *pass_bm,
    lds3_u64 *leaf_bm_flat) {
  const uint32_t tid = threadIdx.x;
  const uint32_t lane = tid & 63;
  const uint32_t wv = tid >> 6;
  const uint32_t iters = (rows + WG - 1) / WG;
  uint8_t cls[8];
  for (uint32_t i = 0; i < n_leaves; i++) {
    cls[i] = leaf_class(cur, plan_leaves[i], bl[i]);
    if (cls[i] != 2) continue;
    leaf_ctx lc = make_leaf_ctx(cur, plan_leaves[i], bl[i], blk_bit);
    const bool slow = lc.slow;
    for (uint32_t it = 0; it < iters; it++) {
      uint32_t rr = it * WG + tid;
      uint32_t r = w0 + rr;
      bool pass = rr < rows;
      if (pass)
        pass = slow ? leaf_match(bv, cur, plan_leaves[i], bl[i], r)
                    : leaf_ctx_match(bv, lc, plan_leaves[i], r);
      uint64_t m = __ballot(pass);
      if (lane == 0)
        leaf_bm_flat[i * (OBX_MAX_BLOCK_ROWS / 64) + it * WAVES + wv] = m;
    }
  }
  __syncthreads();
  for (uint32_t it = 0; it < iters; it++) {
    uint32_t rr = it * WG + tid;
    uint64_t vm = __ballot(rr < rows);
    if (lane == 0) {
      uint64_t stack[8];
      int sp = 0;
      for (uint32_t p = 0; p < ps.n; p++) {
        uint8_t t = ps.t[p];
        if (t < n_leaves) {
          stack[sp++] =
              cls[t] == 2
                  ? leaf_bm_flat[t * (OBX_MAX_BLOCK_ROWS / 64) +
                                 it * WAVES + wv]
                  : cls[t] == 1 ? vm : 0;
        } else if (t == 128) {
          sp--; stack[sp - 1] &= stack[sp];
        } else {
          sp--; stack[sp - 1] |= stack[sp];
        }
      }
      pass_bm[it * WAVES + wv] = stack[0] & vm;
    }
  }
  __syncthreads();
}


template <bool STAGE, bool PROG>
__device__ void scan_filter_agg_body(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr &ph,
    gslot *__restrict__ gtable, unsigned long long *__restrict__ counters,
    uint8_t *lds_blk) {
  __shared__ lds_table tab;
  __shared__ uint64_t pass_bm[OBX_MAX_BLOCK_ROWS / 64];
  /* leaf_bm (phase 1, combine programs only) and row_slot (phases 2-3) are
     live in disjoint phases — share the same 2 KB of LDS */
  __shared__ union {
    uint64_t leaf_bm[8][OBX_MAX_BLOCK_ROWS / 64];
    uint8_t row_slot[OBX_MAX_BLOCK_ROWS];
  } u;
  __shared__ uint8_t cell_slot[64];
  __shared__ unsigned long long wg_passed;

  const uint32_t tid = threadIdx.x;
  const uint32_t lane = tid & 63;
  const uint32_t wv = tid >> 6;


  for (uint32_t s = tid; s < OBX_LTABLE_SLOTS; s += WG) {
    tab.key[s] = OBX_KEY_EMPTY;
    for (uint32_t st = 0; st < OBX_STRIPES; st++) {
      tab.count[s][st] = 0;
      for (uint32_t a = 0; a < ph.n_aggs; a++) {
        uint8_t k = ph.aggs[a].kind;
        tab.cell[s][a][st][0] = (k == 2) ? (unsigned long long)INT64_MAX
                                : (k == 3) ? (unsigned long long)INT64_MIN
                                           : 0ull;
        tab.cell[s][a][st][1] = 0;
      }
    }
  }
  if (tid == 0) wg_passed = 0;
  __syncthreads();

  uint32_t par = 0;
  if (STAGE && OBX_PIPELINE && blockIdx.x < n_blocks)
    stage_issue(buf, blocks[blockIdx.x], lds_blk);
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const dev_block &cur = blocks[b];
    /* constant-result short-circuit: fold the combine program over the
       lowered leaf classes — a FALSE block is skipped before staging (cf.
       the reference's filter constant results / skip-index pruning) */
    uint8_t blk_verdict = 2;
    if (!OBX_PIPELINE || !STAGE) {
      blk_verdict = fold_prog3(ph, cur, plan_leaves,
                               bleaves + (uint64_t)b * ph.n_leaves);
      if (blk_verdict == 0) continue;
    }
    if (STAGE) {
#if OBX_PIPELINE
      stage_wait();
      uint32_t b2 = b + gridDim.x;
      if (b2 < n_blocks)
        stage_issue(buf, blocks[b2],
                    lds_blk + (par ^ 1) * OBX_LDS_STAGE_BYTES);
#else
      __syncthreads(); /* drain previous block's reads */
      stage_issue(buf, cur, lds_blk);
      stage_wait();
#endif
    }
    const uint64_t blk_bit = cur.block_byte * 8;
    blk_view bv;
    bv.base = STAGE ? lds_blk + (OBX_PIPELINE ? par * OBX_LDS_STAGE_BYTES : 0)
                    : buf;
    bv.bit_bias = STAGE ? blk_bit : 0;
    bv.rbase_bit = STAGE ? 0 : blk_bit;
    par ^= 1;

    const uint32_t all_rows = cur.row_count;
    const blk_leaf *bl = bleaves + (uint64_t)b * ph.n_leaves;
    for (uint32_t w0 = 0; w0 < all_rows; w0 += OBX_MAX_BLOCK_ROWS) {
    const uint32_t rows = (all_rows - w0 < OBX_MAX_BLOCK_ROWS)
                              ? all_rows - w0 : OBX_MAX_BLOCK_ROWS;
    const uint32_t iters = (rows + WG - 1) / WG;

    /* ---- phase 1: filter into the LDS pass bitmap, one leaf at a time
       (the reference's per-leaf bitmap combine, ob_pushdown_filter.cpp:1559;
       a single live leaf context keeps uniform state inside the SGPR
       budget). AND-only plans fold progressively; plans with a combine
       program evaluate per-leaf bitmaps then a word-wise postfix pass. ---- */
    if (blk_verdict == 1) {
      for (uint32_t it = 0; it < iters; it++) {
        uint32_t rr = it * WG + tid;
        uint64_t m = __ballot(rr < rows);
        if (lane == 0) pass_bm[it * WAVES + wv] = m;
      }
    } else if constexpr (!PROG) {
      bool first = true;
      for (uint32_t i = 0; i < ph.n_leaves; i++) {
        const blk_leaf lfb = bl[i];
        if (lfb.mode == OBX_LEAF_ALL &&
            !(cur.cols[plan_leaves[i].col].flags & OBX_DF_HAS_EXT))
          continue; /* every row passes this leaf */
        leaf_ctx lc = make_leaf_ctx(cur, plan_leaves[i], bl[i], blk_bit);
        const bool slow = lc.slow;
        for (uint32_t it = 0; it < iters; it++) {
          uint64_t prev = first ? ~0ull : pass_bm[it * WAVES + wv];
          if (!prev) { if (first && lane == 0) pass_bm[it * WAVES + wv] = 0;
                       continue; }
          uint32_t rr = it * WG + tid;
          uint32_t r = w0 + rr;
          bool pass = rr < rows && ((prev >> lane) & 1);
          if (pass)
            pass = slow ? leaf_match(bv, cur, plan_leaves[i], bl[i], r)
                        : leaf_ctx_match(bv, lc, plan_leaves[i], r);
          uint64_t m = __ballot(pass);
          if (lane == 0) pass_bm[it * WAVES + wv] = m;
        }
        first = false;
      }
      if (first) { /* no live leaves: all in-range rows pass */
        for (uint32_t it = 0; it < iters; it++) {
          uint32_t rr = it * WG + tid;
          uint64_t m = __ballot(rr < rows);
          if (lane == 0) pass_bm[it * WAVES + wv] = m;
        }
      }
    } else {
      prog_spec ps;
      ps.n = ph.n_prog;
      for (int p = 0; p < 15; p++) ps.t[p] = ph.prog[p];
      filter_prog_phase(bv, cur, plan_leaves, bl, ph.n_leaves, ps, rows, w0,
                        blk_bit, (lds3_u64 *)pass_bm,
                        (lds3_u64 *)&u.leaf_bm[0][0]);
    }
    /* rows-passed count (lane 0 of each wave over its own words) */
    for (uint32_t it = 0; it < iters; it++) {
      if (lane == 0) {
        uint64_t m = pass_bm[it * WAVES + wv];
        if (m) atomicAdd(&wg_passed, (unsigned long long)__popcll(m));
      }
    }
    __syncthreads();

    /* ---- phase 2: group rows to LDS-table slots (row_slot map).
       Fast path (the reference's storage group-by pushdown design,
       ob_pushdown_aggregate_vec.h:445 / read_reference): when every group
       column is dict-encoded and the ref product is small, rows map to
       cells by packed REFS alone; cell -> slot resolves lazily, once per
       block, via the dict values. NULL group values set flag bits in the
       top key byte (data bytes stay zero). ---- */
    if (ph.n_aggs) {
      col_ctx g0, g1;
      const dev_col *gd0 = nullptr, *gd1 = nullptr;
      if (ph.n_group_cols > 0) {
        gd0 = &cur.cols[ph.need_cols[ph.group_idx[0]]];
        g0 = make_col_ctx(*gd0, blk_bit);
      }
      if (ph.n_group_cols > 1) {
        gd1 = &cur.cols[ph.need_cols[ph.group_idx[1]]];
        g1 = make_col_ctx(*gd1, blk_bit);
      }
      const uint32_t kl0 = ph.group_len[0], kl1 = ph.group_len[1];
      const uint32_t dim0 = (ph.n_group_cols > 0) ? g0.count + 1 : 1;
      const uint32_t dim1 = (ph.n_group_cols > 1) ? g1.count + 1 : 1;
      const bool ref_fast =
          ph.n_group_cols > 0 && g0.kind == 1 &&
          (ph.n_group_cols < 2 || g1.kind == 1) && dim0 * dim1 <= 64;

      if (ref_fast) {
        for (uint32_t i = tid; i < 64; i += WG) cell_slot[i] = 254;
        __syncthreads();
        for (uint32_t it = 0; it < iters; it++) {
          uint32_t rr = it * WG + tid;
          uint32_t r = w0 + rr;
          uint64_t m = pass_bm[it * WAVES + wv];
          bool pass = (m >> lane) & 1;
          if (!m) { if (rr < rows) u.row_slot[rr] = 255; continue; }
          uint8_t slot8 = 255;
          if (pass) {
            uint32_t ref0 = (uint32_t)bit_read_at(
                bv.base, bv.rbase_bit + g0.data_bit + (uint64_t)r * g0.W,
                g0.W);
            if (ref0 > g0.count) ref0 = g0.count; /* nope -> null bucket */
            uint32_t cell = ref0;
            if (ph.n_group_cols > 1) {
              uint32_t ref1 = (uint32_t)bit_read_at(
                  bv.base, bv.rbase_bit + g1.data_bit + (uint64_t)r * g1.W,
                  g1.W);
              if (ref1 > g1.count) ref1 = g1.count;
              cell += ref1 * dim0;
            }
            uint8_t cs = cell_slot[cell];
            if (cs == 254) {
              int s = lds_slot(&tab, build_group_key(bv, ph.n_group_cols,
                                                      gd0, gd1, kl0, kl1,
                                                      r));
              if (s < 0) {
                atomicAdd(&counters[1], 1ull);
                cs = 255;
              } else {
                cs = (uint8_t)s;
              }
              cell_slot[cell] = cs;
            }
            slot8 = cs;
            if (slot8 != 255)
              atomicAdd(&tab.count[slot8][lane & (OBX_STRIPES - 1)], 1ull);
          }
          if (rr < rows) u.row_slot[rr] = slot8;
        }
      } else {
        for (uint32_t it = 0; it < iters; it++) {
          uint32_t rr = it * WG + tid;
          uint32_t r = w0 + rr;
          uint64_t m = pass_bm[it * WAVES + wv];
          bool pass = (m >> lane) & 1;
          if (!m) { if (rr < rows) u.row_slot[rr] = 255; continue; }
          uint8_t slot8 = 255;
          if (pass) {
            int s = lds_slot(&tab, build_group_key(bv, ph.n_group_cols,
                                                   gd0, gd1, kl0, kl1, r));
            if (s < 0) {
              atomicAdd(&counters[1], 1ull);
            } else {
              slot8 = (uint8_t)s;
              atomicAdd(&tab.count[s][lane & (OBX_STRIPES - 1)], 1ull);
            }
          }
          if (rr < rows) u.row_slot[rr] = slot8;
        }
      }
      __syncthreads();

      /* ---- phase 3: one pass per aggregate (ctxs live one at a time).
         The kind switch is hoisted OUT of the row loop; COUNT(*) needs no
         pass (filled from tab.count at flush); PROD2+PROD3 with shared
         inputs run as one fused pass. ---- */
      for (uint32_t a = 0; a < ph.n_aggs; a++) {
        const dev_agg ag = ph.aggs[a];
        if (ag.kind == 0 && ag.ia == 0xFF) continue; /* COUNT(*) at flush */
        const bool fuse_p3 = (ag.kind == 4 && a + 1 < ph.n_aggs &&
                              ph.aggs[a + 1].kind == 5 &&
                              ph.aggs[a + 1].ia == ag.ia &&
                              ph.aggs[a + 1].ib == ag.ib);
        const dev_agg ag2 = fuse_p3 ? ph.aggs[a + 1] : ag;
        col_ctx ca, cb, cc;
        const dev_col *da = nullptr, *db = nullptr, *dc2 = nullptr;
        if (ag.ia != 0xFF) {
          da = &cur.cols[ph.need_cols[ag.ia]];
          ca = make_col_ctx(*da, blk_bit);
        }
        if (ag.ib != 0xFF) {
          db = &cur.cols[ph.need_cols[ag.ib]];
          cb = make_col_ctx(*db, blk_bit);
        }
        uint16_t ic = fuse_p3 ? ag2.ic : ag.ic;
        if (ic != 0xFF && (ag.kind == 5 || fuse_p3)) {
          dc2 = &cur.cols[ph.need_cols[ic]];
          cc = make_col_ctx(*dc2, blk_bit);
        }
        const uint32_t stripe = lane & (OBX_STRIPES - 1);

#define P3_LOOP(...)                                                   \
        for (uint32_t it = 0; it < iters; it++) {                       \
          uint32_t rr = it * WG + tid;                                  \
          uint32_t r = w0 + rr;                                         \
          uint64_t m = pass_bm[it * WAVES + wv];                        \
          if (!m) continue;                                             \
          bool pass = (m >> lane) & 1;                                  \
          uint8_t slot8 = pass && rr < rows ? u.row_slot[rr] : 255;       \
          if (slot8 == 255) continue;                                   \
          int s = slot8;                                                \
          (void)r; __VA_ARGS__                                          \
        }

        switch (ag.kind) {
          case 0: { /* COUNT(col) */
            P3_LOOP({
              bool na;
              (void)((ca.kind == 4) ? col_value2(bv, cur, *da, r, na)
                                    : ctx_value(bv, ca, r, na));
              if (!na)
                atomicAdd(&tab.cell[s][a][stripe][0], 1ull);
            })
            break;
          }
          case 1: { /* SUM */
            P3_LOOP({
              bool na;
              int64_t va = (ca.kind == 4) ? col_value2(bv, cur, *da, r, na)
                                          : ctx_value(bv, ca, r, na);
              if (!na)
                lds_acc_i128(tab.cell[s][a][stripe], i128_from_i64(va));
            })
            break;
          }
          case 2: case 3: { /* MIN / MAX */
            const bool is_min = ag.kind == 2;
            P3_LOOP({
              bool na;
              int64_t va = (ca.kind == 4) ? col_value2(bv, cur, *da, r, na)
                                          : ctx_value(bv, ca, r, na);
              if (!na) {
                cas_minmax(&tab.cell[s][a][stripe][0], va, is_min);
                tab.cell[s][a][stripe][1] = 1;
              }
            })
            break;
          }
          case 4: { /* SUM_PROD2 (optionally fused with a PROD3 mate) */
            if (fuse_p3) {
              P3_LOOP({
                bool na, nb, nc;
                int64_t va = (ca.kind == 4) ? col_value2(bv, cur, *da, r, na)
                                            : ctx_value(bv, ca, r, na);
                int64_t vb = (cb.kind == 4) ? col_value2(bv, cur, *db, r, nb)
                                            : ctx_value(bv, cb, r, nb);
                int64_t vc = (cc.kind == 4) ? col_value2(bv, cur, *dc2, r, nc)
                                            : ctx_value(bv, cc, r, nc);
                if (!na && !nb) {
                  i128v p2 = i128_mul_i64(va, ag.one_b - vb);
                  lds_acc_i128(tab.cell[s][a][stripe], p2);
                  if (!nc)
                    lds_acc_i128(tab.cell[s][a + 1][stripe],
                                 i128_mul_pos_i64(p2, ag2.one_c + vc));
                }
              })
            } else {
              P3_LOOP({
                bool na, nb;
                int64_t va = (ca.kind == 4) ? col_value2(bv, cur, *da, r, na)
                                            : ctx_value(bv, ca, r, na);
                int64_t vb = (cb.kind == 4) ? col_value2(bv, cur, *db, r, nb)
                                            : ctx_value(bv, cb, r, nb);
                if (!na && !nb)
                  lds_acc_i128(tab.cell[s][a][stripe],
                               i128_mul_i64(va, ag.one_b - vb));
              })
            }
            break;
          }
          case 5: { /* SUM_PROD3 standalone */
            P3_LOOP({
              bool na, nb, nc;
              int64_t va = (ca.kind == 4) ? col_value2(bv, cur, *da, r, na)
                                          : ctx_value(bv, ca, r, na);
              int64_t vb = (cb.kind == 4) ? col_value2(bv, cur, *db, r, nb)
                                          : ctx_value(bv, cb, r, nb);
              int64_t vc = (cc.kind == 4) ? col_value2(bv, cur, *dc2, r, nc)
                                          : ctx_value(bv, cc, r, nc);
              if (!na && !nb && !nc)
                lds_acc_i128(tab.cell[s][a][stripe],
                             i128_mul_pos_i64(i128_mul_i64(va,
                                                           ag.one_b - vb),
                                              ag.one_c + vc));
            })
            break;
          }
          case 6: { /* SUM_MUL */
            P3_LOOP({
              bool na, nb;
              int64_t va = (ca.kind == 4) ? col_value2(bv, cur, *da, r, na)
                                          : ctx_value(bv, ca, r, na);
              int64_t vb = (cb.kind == 4) ? col_value2(bv, cur, *db, r, nb)
                                          : ctx_value(bv, cb, r, nb);
              if (!na && !nb)
                lds_acc_i128(tab.cell[s][a][stripe], i128_mul_i64(va, vb));
            })
            break;
          }
          default: break;
        }
#undef P3_LOOP
        if (fuse_p3) a++; /* consumed the PROD3 mate */
      }
    }
    __syncthreads(); /* pass_bm reuse across windows */
    } /* window loop */
  }
  __syncthreads();

  /* flush LDS table to the global table (merge lane stripes) */
  if (tid == 0 && wg_passed) atomicAdd(&counters[0], wg_passed);
  for (uint32_t s = tid; s < OBX_LTABLE_SLOTS; s += WG) {
    uint64_t key = tab.key[s];
    if (key == OBX_KEY_EMPTY) continue;
    uint32_t idx = (uint32_t)((key * 0x9E3779B97F4A7C15ull) >> (64 - OBX_GTABLE_SHIFT)) &
                   (OBX_GTABLE_SLOTS - 1);
    for (int probe = 0; probe < OBX_GTABLE_SLOTS; probe++) {
      unsigned long long cur_k = atomicCAS(&gtable[idx].key, OBX_KEY_EMPTY,
                                           (unsigned long long)key);
      if (cur_k == OBX_KEY_EMPTY || cur_k == key) break;
      idx = (idx + 1) & (OBX_GTABLE_SLOTS - 1);
    }
    unsigned long long cnt = 0;
    for (uint32_t st = 0; st < OBX_STRIPES; st++) cnt += tab.count[s][st];
    atomicAdd(&gtable[idx].count, cnt);
    for (uint32_t a = 0; a < ph.n_aggs; a++) {
      uint8_t kind = ph.aggs[a].kind;
      if (kind == 0 && ph.aggs[a].ia == 0xFF) {
        /* COUNT(*): equals the group's row count */
        g_acc_i128(gtable[idx].cells[a], cnt, 0);
      } else if (kind == 2 || kind == 3) {
        for (uint32_t st = 0; st < OBX_STRIPES; st++) {
          if (tab.cell[s][a][st][1]) {
            cas_minmax(&gtable[idx].cells[a][0],
                       (int64_t)tab.cell[s][a][st][0], kind == 2);
            gtable[idx].cells[a][1] = 1;
          }
        }
      } else {
        /* sum the 4 striped int128 partials, then one 256-bit accumulate */
        uint64_t lo = 0, hi = 0;
        for (uint32_t st = 0; st < OBX_STRIPES; st++) {
          uint64_t l = tab.cell[s][a][st][0];
          uint64_t nlo = lo + l;
          hi += tab.cell[s][a][st][1] + (nlo < l);
          lo = nlo;
        }
        g_acc_i128(gtable[idx].cells[a], lo, hi);
      }
    }
  }
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_scan_filter_agg(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    gslot *__restrict__ gtable, unsigned long long *__restrict__ counters) {
  scan_filter_agg_body<false, false>(buf, blocks, n_blocks, plan_leaves,
                                     bleaves, ph, gtable, counters, nullptr);
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_scan_filter_agg_prog(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    gslot *__restrict__ gtable, unsigned long long *__restrict__ counters) {
  scan_filter_agg_body<false, true>(buf, blocks, n_blocks, plan_leaves,
                                    bleaves, ph, gtable, counters, nullptr);
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_scan_filter_agg_lds(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    gslot *__restrict__ gtable, unsigned long long *__restrict__ counters) {
  __shared__ uint8_t lds_blk[(1 + OBX_PIPELINE) * OBX_LDS_STAGE_BYTES + 32];
  scan_filter_agg_body<true, false>(buf, blocks, n_blocks, plan_leaves,
                                    bleaves, ph, gtable, counters, lds_blk);
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_scan_filter_agg_prog_lds(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    gslot *__restrict__ gtable, unsigned long long *__restrict__ counters) {
  __shared__ uint8_t lds_blk[(1 + OBX_PIPELINE) * OBX_LDS_STAGE_BYTES + 32];
  scan_filter_agg_body<true, true>(buf, blocks, n_blocks, plan_leaves,
                                   bleaves, ph, gtable, counters, lds_blk);
}

/* ---------------- filter-only kernel (bitmap + selection vectors) ------- */
template <bool STAGE, bool PROG>
__device__ void filter_body(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr &ph,
    uint64_t *__restrict__ bitmap, int32_t *__restrict__ row_ids,
    uint32_t *__restrict__ blk_counts,
    unsigned long long *__restrict__ counters, uint8_t *lds_blk) {
  __shared__ uint32_t wv_cnt[WAVES];
  __shared__ uint32_t wv_scan[WAVES];
  __shared__ uint32_t blk_written;
  __shared__ unsigned long long wg_passed;
  __shared__ uint64_t pass_bm[OBX_MAX_BLOCK_ROWS / 64];
  __shared__ uint64_t leaf_bm[8][OBX_MAX_BLOCK_ROWS / 64];

  const uint32_t tid = threadIdx.x;
  const uint32_t lane = tid & 63;
  const uint32_t wv = tid >> 6;
  if (tid == 0) wg_passed = 0;
  __syncthreads();

  uint32_t par = 0;
  if (STAGE && OBX_PIPELINE && blockIdx.x < n_blocks)
    stage_issue(buf, blocks[blockIdx.x], lds_blk);
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const dev_block &cur = blocks[b];
    uint8_t blk_verdict = 2;
    if (!OBX_PIPELINE || !STAGE) {
      /* constant-result short-circuit (see scan_filter_agg_body) */
      blk_verdict = fold_prog3(ph, cur, plan_leaves,
                               bleaves + (uint64_t)b * ph.n_leaves);
      if (blk_verdict == 0) {
        if (row_ids && tid == 0) blk_counts[b] = 0;
        continue; /* bitmap is pre-zeroed */
      }
    }
    if (tid == 0) blk_written = 0;
    if (STAGE) {
#if OBX_PIPELINE
      stage_wait();
      uint32_t b2 = b + gridDim.x;
      if (b2 < n_blocks)
        stage_issue(buf, blocks[b2],
                    lds_blk + (par ^ 1) * OBX_LDS_STAGE_BYTES);
#else
      __syncthreads();
      stage_issue(buf, cur, lds_blk);
      stage_wait();
#endif
    } else {
      __syncthreads(); /* blk_written reset */
    }
    const uint64_t blk_bit = cur.block_byte * 8;
    blk_view bv;
    bv.base = STAGE ? lds_blk + (OBX_PIPELINE ? par * OBX_LDS_STAGE_BYTES : 0)
                    : buf;
    bv.bit_bias = STAGE ? blk_bit : 0;
    bv.rbase_bit = STAGE ? 0 : blk_bit;
    par ^= 1;

    const uint32_t all_rows = cur.row_count;
    const uint64_t row_start = dev_block_row_start(&cur);
    const blk_leaf *bl = bleaves + (uint64_t)b * ph.n_leaves;
    for (uint32_t w0 = 0; w0 < all_rows; w0 += OBX_MAX_BLOCK_ROWS) {
    const uint32_t rows = (all_rows - w0 < OBX_MAX_BLOCK_ROWS)
                              ? all_rows - w0 : OBX_MAX_BLOCK_ROWS;
    const uint32_t iters = (rows + WG - 1) / WG;

    /* leaf-by-leaf fission into the pass bitmap (one live context);
       combine programs evaluate per-leaf bitmaps then a word-wise postfix */
    if (blk_verdict == 1) {
      for (uint32_t it = 0; it < iters; it++) {
        uint32_t rr = it * WG + tid;
        uint64_t m = __ballot(rr < rows);
        if (lane == 0) pass_bm[it * WAVES + wv] = m;
      }
    } else if constexpr (!PROG) {
      bool first = true;
      for (uint32_t i = 0; i < ph.n_leaves; i++) {
        const blk_leaf lfb = bl[i];
        if (lfb.mode == OBX_LEAF_ALL &&
            !(cur.cols[plan_leaves[i].col].flags & OBX_DF_HAS_EXT))
          continue;
        leaf_ctx lc = make_leaf_ctx(cur, plan_leaves[i], bl[i], blk_bit);
        const bool slow = lc.slow;
        for (uint32_t it = 0; it < iters; it++) {
          uint64_t prev = first ? ~0ull : pass_bm[it * WAVES + wv];
          if (!prev) { if (first && lane == 0) pass_bm[it * WAVES + wv] = 0;
                       continue; }
          uint32_t rr = it * WG + tid;
          uint32_t r = w0 + rr;
          bool pass = rr < rows && ((prev >> lane) & 1);
          if (pass)
            pass = slow ? leaf_match(bv, cur, plan_leaves[i], bl[i], r)
                        : leaf_ctx_match(bv, lc, plan_leaves[i], r);
          uint64_t m = __ballot(pass);
          if (lane == 0) pass_bm[it * WAVES + wv] = m;
        }
        first = false;
      }
      if (first) {
        for (uint32_t it = 0; it < iters; it++) {
          uint32_t rr = it * WG + tid;
          uint64_t m = __ballot(rr < rows);
          if (lane == 0) pass_bm[it * WAVES + wv] = m;
        }
      }
    } else {
      prog_spec ps;
      ps.n = ph.n_prog;
      for (int p = 0; p < 15; p++) ps.t[p] = ph.prog[p];
      filter_prog_phase(bv, cur, plan_leaves, bl, ph.n_leaves, ps, rows, w0,
                        blk_bit, (lds3_u64 *)pass_bm,
                        (lds3_u64 *)&leaf_bm[0][0]);
    }

    /* output: global bitmap words, pass counts, selection vectors */
    for (uint32_t it = 0; it < iters; it++) {
      uint64_t m = pass_bm[it * WAVES + wv];
      if (bitmap && lane == 0 && m) {
        uint64_t gbit = row_start + w0 + (uint64_t)it * WG +
                        (uint64_t)wv * 64;
        uint64_t widx = gbit >> 6;
        uint32_t sh = (uint32_t)(gbit & 63);
        atomicOr((unsigned long long *)&bitmap[widx],
                 (unsigned long long)(m << sh));
        if (sh && (m >> (64 - sh)))
          atomicOr((unsigned long long *)&bitmap[widx + 1],
                   (unsigned long long)(m >> (64 - sh)));
      }
      if (row_ids) {
        if (lane == 0) wv_cnt[wv] = (uint32_t)__popcll(m);
        __syncthreads();
        if (tid == 0) {
          uint32_t acc = blk_written;
          for (uint32_t w = 0; w < WAVES; w++) {
            wv_scan[w] = acc;
            acc += wv_cnt[w];
          }
          blk_written = acc;
        }
        __syncthreads();
        uint32_t r = w0 + it * WG + tid;
        bool pass = (m >> lane) & 1;
        if (pass) {
          uint32_t below = __builtin_amdgcn_mbcnt_lo((uint32_t)m, 0);
          below = __builtin_amdgcn_mbcnt_hi((uint32_t)(m >> 32), below);
          row_ids[row_start + wv_scan[wv] + below] = (int32_t)r;
        }
        __syncthreads();
      } else if (lane == 0 && m) {
        atomicAdd(&wg_passed, (unsigned long long)__popcll(m));
      }
    }
    __syncthreads(); /* pass_bm reuse across windows */
    } /* window loop */
    if (row_ids) {
      __syncthreads();
      if (tid == 0) {
        blk_counts[b] = blk_written;
        atomicAdd(&wg_passed, (unsigned long long)blk_written);
      }
    }
    __syncthreads();
  }
  if (tid == 0 && wg_passed) atomicAdd(&counters[0], wg_passed);
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_filter(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    uint64_t *__restrict__ bitmap, int32_t *__restrict__ row_ids,
    uint32_t *__restrict__ blk_counts,
    unsigned long long *__restrict__ counters) {
  filter_body<false, false>(buf, blocks, n_blocks, plan_leaves, bleaves, ph,
                            bitmap, row_ids, blk_counts, counters, nullptr);
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_filter_prog(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    uint64_t *__restrict__ bitmap, int32_t *__restrict__ row_ids,
    uint32_t *__restrict__ blk_counts,
    unsigned long long *__restrict__ counters) {
  filter_body<false, true>(buf, blocks, n_blocks, plan_leaves, bleaves, ph,
                           bitmap, row_ids, blk_counts, counters, nullptr);
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_filter_lds(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    uint64_t *__restrict__ bitmap, int32_t *__restrict__ row_ids,
    uint32_t *__restrict__ blk_counts,
    unsigned long long *__restrict__ counters) {
  __shared__ uint8_t lds_blk[(1 + OBX_PIPELINE) * OBX_LDS_STAGE_BYTES + 32];
  filter_body<true, false>(buf, blocks, n_blocks, plan_leaves, bleaves, ph,
                           bitmap, row_ids, blk_counts, counters, lds_blk);
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_filter_prog_lds(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    uint64_t *__restrict__ bitmap, int32_t *__restrict__ row_ids,
    uint32_t *__restrict__ blk_counts,
    unsigned long long *__restrict__ counters) {
  __shared__ uint8_t lds_blk[(1 + OBX_PIPELINE) * OBX_LDS_STAGE_BYTES + 32];
  filter_body<true, true>(buf, blocks, n_blocks, plan_leaves, bleaves, ph,
                          bitmap, row_ids, blk_counts, counters, lds_blk);
}

/* ---------------- decode kernel (get_rows equivalent, parity) ----------- */
extern "C" __global__ __launch_bounds__(WG, 2) void k_decode(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, uint32_t col, uint32_t datum_len,
    uint8_t *__restrict__ out, uint8_t *__restrict__ out_null) {
  const uint32_t tid = threadIdx.x;
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const dev_block &cur = blocks[b];
    blk_view bv; bv.base = buf; bv.bit_bias = 0; bv.rbase_bit = 0;
    const uint32_t rows = cur.row_count;
    const uint64_t row_start = dev_block_row_start(&cur);
    for (uint32_t r = tid; r < rows; r += WG) {
      bool isn;
      int64_t v = col_value2(bv, cur, cur.cols[col], r, isn);
      uint64_t uv = isn ? 0 : (uint64_t)v;
      uint8_t *dst = out + (row_start + r) * datum_len;
      if (datum_len == 8)       /* aligned: hipMalloc base + 8-B stride */
        *(uint64_t *)dst = uv;
      else if (datum_len == 4)
        *(uint32_t *)dst = (uint32_t)uv;
      else
        for (uint32_t i = 0; i < datum_len; i++)
          dst[i] = (uint8_t)(uv >> (i * 8));
      if (out_null) out_null[row_start + r] = isn ? 1 : 0;
    }
  }
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_decode_lds(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, uint32_t col, uint32_t datum_len,
    uint8_t *__restrict__ out, uint8_t *__restrict__ out_null) {
  /* staged get_rows: per-lane bit-granular reads hit LDS instead of L2
     (the global-read variant measured ~0.8 TB/s; staging reads the
     encoded bytes once, coalesced) */
  __shared__ uint8_t lds_blk[OBX_LDS_STAGE_BYTES + 32];
  const uint32_t tid = threadIdx.x;
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const dev_block &cur = blocks[b];
    __syncthreads(); /* drain previous block's LDS reads */
    stage_issue(buf, cur, lds_blk);
    stage_wait();
    blk_view bv;
    bv.base = lds_blk;
    bv.bit_bias = cur.block_byte * 8;
    bv.rbase_bit = 0;
    const uint32_t rows = cur.row_count;
    const uint64_t row_start = dev_block_row_start(&cur);
    for (uint32_t r = tid; r < rows; r += WG) {
      bool isn;
      int64_t v = col_value2(bv, cur, cur.cols[col], r, isn);
      uint64_t uv = isn ? 0 : (uint64_t)v;
      uint8_t *dst = out + (row_start + r) * datum_len;
      if (datum_len == 8)       /* aligned: hipMalloc base + 8-B stride */
        *(uint64_t *)dst = uv;
      else if (datum_len == 4)
        *(uint32_t *)dst = (uint32_t)uv;
      else
        for (uint32_t i = 0; i < datum_len; i++)
          dst[i] = (uint8_t)(uv >> (i * 8));
      if (out_null) out_null[row_start + r] = isn ? 1 : 0;
    }
  }
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_decode_multi(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const uint16_t *__restrict__ proj,
    const uint8_t *__restrict__ lens, uint32_t n_proj,
    uint8_t *const *__restrict__ outs, uint8_t *const *__restrict__ onulls) {
  /* one stage per block, ALL projected columns decoded from LDS — the
     per-column variant re-reads every block once per column */
  __shared__ uint8_t lds_blk[OBX_LDS_STAGE_BYTES + 32];
  const uint32_t tid = threadIdx.x;
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const dev_block &cur = blocks[b];
    __syncthreads(); /* drain previous block's LDS reads */
    stage_issue(buf, cur, lds_blk);
    stage_wait();
    blk_view bv;
    bv.base = lds_blk;
    bv.bit_bias = cur.block_byte * 8;
    bv.rbase_bit = 0;
    const uint32_t rows = cur.row_count;
    const uint64_t row_start = dev_block_row_start(&cur);
    for (uint32_t p = 0; p < n_proj; p++) {
      const uint32_t col = proj[p];
      const uint32_t datum_len = lens[p];
      uint8_t *out = outs[p];
      uint8_t *out_null = onulls ? onulls[p] : nullptr;
      for (uint32_t r = tid; r < rows; r += WG) {
        bool isn;
        int64_t v = col_value2(bv, cur, cur.cols[col], r, isn);
        uint64_t uv = isn ? 0 : (uint64_t)v;
        uint8_t *dst = out + (row_start + r) * datum_len;
        if (datum_len == 8)
          *(uint64_t *)dst = uv;
        else if (datum_len == 4)
          *(uint32_t *)dst = (uint32_t)uv;
        else
          for (uint32_t i = 0; i < datum_len; i++)
            dst[i] = (uint8_t)(uv >> (i * 8));
        if (out_null) out_null[row_start + r] = isn ? 1 : 0;
      }
    }
  }
}

/* ---------------- device-side filter lowering ---------------------------
 * One thread per (block, leaf): translate the leaf into the block's packed
 * domain (the reference evaluates dict-domain filters once per dict entry,
 * ob_dict_decoder.cpp:810-886,1481-1561; and maps raw compares to its
 * fast SIMD paths, ob_raw_decoder.cpp:707-790). */
__device__ __forceinline__ void lower_range(__int128 D, __int128 D2,
                                            uint8_t op, uint64_t dmax,
                                            blk_leaf &out) {
  /* unsigned packed domain: v in [0, dmax]; D/D2 are the operands already
     mapped into that domain (may lie outside [0, dmax]) */
  out.invert = 0;
  switch (op) {
    case 0: /* EQ */
      if (D < 0 || D > (__int128)dmax) { out.mode = OBX_LEAF_NONE; return; }
      out.mode = OBX_LEAF_RANGE; out.lo = (uint64_t)D; out.hi = (uint64_t)D;
      return;
    case 5: /* NE */
      if (D < 0 || D > (__int128)dmax) { out.mode = OBX_LEAF_ALL; return; }
      out.mode = OBX_LEAF_RANGE; out.lo = (uint64_t)D; out.hi = (uint64_t)D;
      out.invert = 1;
      return;
    case 1: /* LE */
      if (D < 0) { out.mode = OBX_LEAF_NONE; return; }
      if (D >= (__int128)dmax) { out.mode = OBX_LEAF_ALL; return; }
      out.mode = OBX_LEAF_RANGE; out.lo = 0; out.hi = (uint64_t)D;
      return;
    case 2: /* LT */
      if (D <= 0) { out.mode = OBX_LEAF_NONE; return; }
      if (D > (__int128)dmax) { out.mode = OBX_LEAF_ALL; return; }
      out.mode = OBX_LEAF_RANGE; out.lo = 0; out.hi = (uint64_t)(D - 1);
      return;
    case 3: /* GE */
      if (D <= 0) { out.mode = OBX_LEAF_ALL; return; }
      if (D > (__int128)dmax) { out.mode = OBX_LEAF_NONE; return; }
      out.mode = OBX_LEAF_RANGE; out.lo = (uint64_t)D; out.hi = dmax;
      return;
    case 4: /* GT */
      if (D < 0) { out.mode = OBX_LEAF_ALL; return; }
      if (D >= (__int128)dmax) { out.mode = OBX_LEAF_NONE; return; }
      out.mode = OBX_LEAF_RANGE; out.lo = (uint64_t)(D + 1); out.hi = dmax;
      return;
    case 6: { /* BT */
      __int128 lo = D < 0 ? 0 : D;
      __int128 hi = D2 > (__int128)dmax ? (__int128)dmax : D2;
      if (lo > hi) { out.mode = OBX_LEAF_NONE; return; }
      out.mode = OBX_LEAF_RANGE; out.lo = (uint64_t)lo; out.hi = (uint64_t)hi;
      return;
    }
    default:
      out.mode = OBX_LEAF_VALUE;
      return;
  }
}

extern "C" __global__ void k_lower_leaves(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ pl, uint32_t n_leaves,
    uint32_t n_cols, const int64_t *__restrict__ minmax,
    blk_leaf *__restrict__ out) {
  uint64_t idx = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t total = (uint64_t)n_blocks * n_leaves;
  if (idx >= total) return;
  uint32_t b = (uint32_t)(idx / n_leaves);
  uint32_t i = (uint32_t)(idx % n_leaves);
  const dev_leaf &lf = pl[i];
  const dev_col &c = blocks[b].cols[lf.col];
  blk_view bv; bv.base = buf; bv.bit_bias = 0; bv.rbase_bit = 0;
  blk_leaf o;
  o.mask = 0; o.lo = 0; o.hi = 0; o.mode = OBX_LEAF_VALUE; o.invert = 0;

  /* black (generic-expression) leaves: the filter-on-dict trick — a
     single-column program over a small dict evaluates ONCE PER ENTRY
     into a ref mask (ob_dict_decoder.cpp:1481-1561); a const column
     folds to ALL/NONE; anything else evaluates per row (VALUE mode,
     black_match). */
  if (lf.op == 10) {
    const dev_col &bc = blocks[b].cols[lf.bcols[0]];
    if (lf.n_bcols == 1 &&
        (bc.enc == OBX_D_DICT || bc.enc == OBX_D_RLE) && bc.count < 64) {
      for (uint32_t e = 0; e < bc.count; e++) {
        int64_t v = dict_entry(bv, bc, e);
        bool nu0 = false;
        if (black_eval_vals(lf, &v, &nu0)) o.mask |= 1ull << e;
      }
      o.mode = (o.mask == 0) ? OBX_LEAF_NONE : OBX_LEAF_REF_MASK;
    } else if (lf.n_bcols == 1 && bc.enc == OBX_D_CONST && bc.runs == 0) {
      int64_t v = bc.base;
      bool nu0 = (bc.count == 0);
      o.mode = black_eval_vals(lf, &v, &nu0) ? OBX_LEAF_ALL : OBX_LEAF_NONE;
    } else {
      o.mode = OBX_LEAF_VALUE;
    }
    out[idx] = o;
    return;
  }

  /* NOT const-with-exceptions: REF_MASK row evaluation reads a per-row
     ref (packed stream or RLE runs), which CONST does not have — its
     refs live in the exception list, so it takes the VALUE slow path
     (col_value2 walks the exceptions). */
  const bool is_dict = (c.enc == OBX_D_DICT || c.enc == OBX_D_RLE);
  if (is_dict && c.count < 64) {
    if (lf.op == 8) {
      o.mask = 1ull << c.count;
    } else if (lf.op == 9) {
      o.mask = (c.count == 0) ? 0 : ((1ull << c.count) - 1);
    } else {
      for (uint32_t e = 0; e < c.count; e++) {
        int64_t v = dict_entry(bv, c, e);
        if (leaf_value_match(lf, v, false)) o.mask |= 1ull << e;
      }
    }
    o.mode = (o.mask == 0) ? OBX_LEAF_NONE : OBX_LEAF_REF_MASK;
  } else if (c.enc == OBX_D_RAW && (c.flags & OBX_DF_BITPACK) && lf.op <= 6) {
    uint64_t dmax = (c.width >= 64) ? ~0ull : ((1ull << c.width) - 1);
    lower_range((__int128)lf.vlo, (__int128)lf.vhi, lf.op, dmax, o);
  } else if (c.enc == OBX_D_RAW && !(c.flags & OBX_DF_BITPACK) &&
             !(c.flags & OBX_DF_STRING) && lf.op <= 6 && lf.op != 7 /* IN */) {
    /* fixed-width RAW numeric: k-byte domain. Signed iff the sign bit lives
       inside the stored bytes (INT class with width==tss, or DECIMAL);
       order-map signed domains to unsigned with an xor bias. */
    uint32_t kbits = (uint32_t)c.width * 8;
    uint64_t dmax = (kbits >= 64) ? ~0ull : ((1ull << kbits) - 1);
    bool signed_dom = ((c.flags & OBX_DF_SIGNED) && c.width >= c.tss) ||
                      (!(c.flags & OBX_DF_SIGNED) && c.width == 8);
    if (signed_dom) {
      uint64_t bias = 1ull << (kbits - 1);
      __int128 smin = -(__int128)bias, smax = (__int128)bias - 1;
      auto biasmap = [&](int64_t x) -> __int128 {
        __int128 v = x;
        if (v < smin) return (__int128)-1;          /* below domain */
        if (v > smax) return (__int128)dmax + 1;    /* above domain */
        return v + (__int128)bias;                  /* in [0, dmax] */
      };
      lower_range(biasmap(lf.vlo), biasmap(lf.vhi), lf.op, dmax, o);
      if (o.mode == OBX_LEAF_RANGE) o.mask = bias;
    } else {
      lower_range((__int128)lf.vlo, (__int128)lf.vhi, lf.op, dmax, o);
    }
  } else if (c.enc == OBX_D_INTDIFF && lf.op <= 6) {
    uint32_t kbits = (c.flags & OBX_DF_BITPACK) ? c.width : c.width * 8;
    uint64_t dmax = (kbits >= 64) ? ~0ull : ((1ull << kbits) - 1);
    lower_range((__int128)lf.vlo - c.base, (__int128)lf.vhi - c.base, lf.op,
                dmax, o);
  } else if ((c.enc == OBX_D_RAW || c.enc == OBX_D_INTDIFF) &&
             (lf.op == 8 || lf.op == 9)) {
    if (c.flags & OBX_DF_HAS_EXT) {
      o.mode = OBX_LEAF_NULL;
      o.invert = (lf.op == 9);
    } else {
      o.mode = (lf.op == 8) ? OBX_LEAF_NONE : OBX_LEAF_ALL;
    }
  } else if (c.enc == OBX_D_CONST && c.runs == 0) {
    bool isn = (c.count == 0);
    bool m = leaf_value_match(lf, c.base, isn);
    o.mode = m ? OBX_LEAF_ALL : OBX_LEAF_NONE;
  } else {
    o.mode = OBX_LEAF_VALUE;
  }

  /* Stored min/max skip-index refinement (the rebuild of the reference's
   * ObSSTableIndexFilter micro-block pruning,
   * /root/reference/src/storage/access/ob_sstable_index_filter.cpp: skip
   * a micro block whose [min,max] cannot satisfy the predicate, or mark
   * it all-pass). The per-(block,col) bounds are captured once at load
   * by k_col_minmax (the analogue of the skip index written at encode
   * time); strings are excluded (VALUE-mode compares map char through
   * char_key order, which the raw-LE bounds do not follow). */
  if (minmax && lf.op <= 6 && !(c.flags & OBX_DF_STRING) &&
      (o.mode == OBX_LEAF_RANGE || o.mode == OBX_LEAF_VALUE)) {
    int64_t mn = minmax[2 * ((uint64_t)b * n_cols + lf.col)];
    int64_t mx = minmax[2 * ((uint64_t)b * n_cols + lf.col) + 1];
    if (!(mn == INT64_MIN && mx == INT64_MAX)) { /* known bounds */
      bool none = false, all = false;
      if (mn > mx) { /* no non-null values in the block */
        none = true;
      } else {
        const int64_t lo = lf.vlo, hi = lf.vhi;
        switch (lf.op) {
          case 0: none = lo < mn || lo > mx; all = (mn == mx && mn == lo);
                  break;
          case 1: none = mn > lo; all = mx <= lo; break; /* LE */
          case 2: none = mn >= lo; all = mx < lo; break; /* LT */
          case 3: none = mx < lo; all = mn >= lo; break; /* GE */
          case 4: none = mx <= lo; all = mn > lo; break; /* GT */
          case 5: none = (mn == mx && mn == lo); all = lo < mn || lo > mx;
                  break;
          case 6: none = mx < lo || mn > hi; all = mn >= lo && mx <= hi;
                  break;
        }
      }
      if (none) o.mode = OBX_LEAF_NONE;
      else if (all) o.mode = OBX_LEAF_ALL; /* non-null rows all pass; ext
                                              nulls still excluded by the
                                              kernels' ALL handling */
    }
  }
  out[idx] = o;
}

/* Per-(block,column) min/max of the NON-NULL decoded values, captured once
 * at load time — the engine's stored skip index (the reference writes
 * ObSkipIndex min/max rows at encode time, storage/blocksstable/index_block;
 * our container has no index block, so the bounds are derived from the
 * encoded data at load, outside every timed region). One wave per
 * (block,col). Output: out[2*(b*n_cols+c)] = {min, max};
 * {INT64_MIN, INT64_MAX} = unknown (strings/span encodings);
 * min > max = no non-null values. */
extern "C" __global__ void k_col_minmax(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, uint32_t n_cols, int64_t *__restrict__ out) {
  const uint32_t lane = threadIdx.x & 63;
  const uint32_t wv = threadIdx.x >> 6;
  uint64_t total = (uint64_t)n_blocks * n_cols;
  for (uint64_t wid = (uint64_t)blockIdx.x * WAVES + wv; wid < total;
       wid += (uint64_t)gridDim.x * WAVES) {
    uint32_t b = (uint32_t)(wid / n_cols), c = (uint32_t)(wid % n_cols);
    const dev_block &blk = blocks[b];
    const dev_col &dc = blk.cols[c];
    blk_view bv;
    bv.base = buf;
    bv.bit_bias = 0;
    bv.rbase_bit = 0;
    int64_t mn = INT64_MAX, mx = INT64_MIN;
    bool known = true;
    if (dc.flags & OBX_DF_STRING) {
      known = false;
    } else {
      switch (dc.enc) {
        case OBX_D_DICT:
        case OBX_D_RLE:
          for (uint32_t e = lane; e < dc.count; e += 64) {
            int64_t v = dict_entry(bv, dc, e);
            if (v < mn) mn = v;
            if (v > mx) mx = v;
          }
          break;
        case OBX_D_CONST:
          if (dc.runs == 0) {
            if (lane == 0 && dc.count) mn = mx = dc.base;
          } else {
            for (uint32_t e = lane; e < dc.count; e += 64) {
              int64_t v = dict_entry(bv, dc, e);
              if (v < mn) mn = v;
              if (v > mx) mx = v;
            }
          }
          break;
        case OBX_D_RAW:
        case OBX_D_INTDIFF:
          for (uint32_t r = lane; r < blk.row_count; r += 64) {
            bool isn;
            int64_t v = col_value(bv, dc, r, isn);
            if (!isn) {
              if (v < mn) mn = v;
              if (v > mx) mx = v;
            }
          }
          break;
        default:
          known = false;
          break;
      }
    }
    if (!known) {
      mn = INT64_MIN;
      mx = INT64_MAX;
    }
    for (int off = 32; off > 0; off >>= 1) {
      int64_t omn = (int64_t)shflxor64((uint64_t)mn, off);
      int64_t omx = (int64_t)shflxor64((uint64_t)mx, off);
      if (known) { /* unknown is wave-uniform (depends only on dc) */
        if (omn < mn) mn = omn;
        if (omx > mx) mx = omx;
      }
    }
    if (lane == 0) {
      out[2 * wid] = mn;
      out[2 * wid + 1] = mx;
    }
  }
}

/* ======================================================================
 * Pass-pipeline aggregation (round-1 final design).
 *
 * The one-kernel fused design pays SGPR-spill scratch traffic and staging
 * barriers (profiles/r01_q1_sf100.md). The query instead runs as a short
 * kernel DAG — each pass small and register-lean, reading only its own
 * column streams directly from HBM (consecutive rows of one column are
 * contiguous -> coalesced; dict payloads stay hot in L1/L2):
 *
 *   k_filter(_lds)  -> survivor bitmap (1 bit/row)   [existing kernel]
 *   k_group_pass    -> row_slot[] (u8/row), global group keys + counts
 *   k_agg_pass x P  -> one aggregate (or a fused PROD2+PROD3 pair) per
 *                      pass; per-slot LDS cells, one global flush per WG
 *
 * Kernel boundaries are the inter-pass synchronization (~1.5 us each).
 * Slots are indices into the global group table (gslot); row_slot stores
 * them as u8 (255 = filtered out or table overflow; overflow counted).
 * ====================================================================== */

/* find-or-insert a key in the global group table; returns slot or -1 */
__device__ __forceinline__ int g_slot(gslot *gtable, uint64_t key) {
  uint32_t idx = (uint32_t)((key * 0x9E3779B97F4A7C15ull) >> (64 - OBX_GTABLE_SHIFT)) &
                 (OBX_GTABLE_SLOTS - 1);
  for (int probe = 0; probe < OBX_GTABLE_SLOTS; probe++) {
    unsigned long long k = __hip_atomic_load(
        &gtable[idx].key, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (k == key) return (int)idx;
    if (k == OBX_KEY_EMPTY) {
      unsigned long long cur = atomicCAS(&gtable[idx].key, OBX_KEY_EMPTY,
                                         (unsigned long long)key);
      if (cur == OBX_KEY_EMPTY || cur == key) return (int)idx;
    }
    idx = (idx + 1) & (OBX_GTABLE_SLOTS - 1);
  }
  return -1;
}

/* this row's bit from the global survivor bitmap */
__device__ __forceinline__ bool pass_bit(const uint64_t *__restrict__ bitmap,
                                         uint64_t grow) {
  return (bitmap[grow >> 6] >> (grow & 63)) & 1;
}

/* ---- group pass: map each surviving row to a global-table slot ---- */
extern "C" __global__ __launch_bounds__(WG, 2) void k_group_pass(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_plan_hdr ph,
    const uint64_t *__restrict__ bitmap, uint8_t *__restrict__ row_slot,
    gslot *__restrict__ gtable, unsigned long long *__restrict__ counters) {
  __shared__ uint8_t cell_slot[64];
  __shared__ unsigned long long cnt[OBX_GTABLE_SLOTS][4]; /* lane-striped */
  const uint32_t tid = threadIdx.x;
  const uint32_t lane = tid & 63;

  for (uint32_t i = tid; i < OBX_GTABLE_SLOTS; i += WG) {
    cnt[i][0] = 0; cnt[i][1] = 0; cnt[i][2] = 0; cnt[i][3] = 0;
  }
  __syncthreads();

  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const dev_block &cur = blocks[b];
    const uint64_t blk_bit = cur.block_byte * 8;
    blk_view bv;
    bv.base = buf; bv.bit_bias = 0; bv.rbase_bit = blk_bit;
    const uint32_t rows = cur.row_count;
    const uint64_t row_start = dev_block_row_start(&cur);

    col_ctx g0, g1;
    const dev_col *gd0 = nullptr, *gd1 = nullptr;
    if (ph.n_group_cols > 0) {
      gd0 = &cur.cols[ph.need_cols[ph.group_idx[0]]];
      g0 = make_col_ctx(*gd0, blk_bit);
    }
    if (ph.n_group_cols > 1) {
      gd1 = &cur.cols[ph.need_cols[ph.group_idx[1]]];
      g1 = make_col_ctx(*gd1, blk_bit);
    }
    const uint32_t kl0 = ph.group_len[0], kl1 = ph.group_len[1];
    const uint32_t dim0 = (ph.n_group_cols > 0) ? g0.count + 1 : 1;
    const uint32_t dim1 = (ph.n_group_cols > 1) ? g1.count + 1 : 1;
    const bool ref_fast = ph.n_group_cols > 0 && g0.kind == 1 &&
                          (ph.n_group_cols < 2 || g1.kind == 1) &&
                          dim0 * dim1 <= 64;
    if (ref_fast) {
      __syncthreads(); /* previous block's cell reads drain */
      for (uint32_t i = tid; i < 64; i += WG) cell_slot[i] = 254;
      __syncthreads();
    }

    const uint32_t iters = (rows + WG - 1) / WG;
    for (uint32_t it = 0; it < iters; it++) {
      uint32_t r = it * WG + tid;
      bool in = r < rows;
      bool pass = in && pass_bit(bitmap, row_start + r);
      uint8_t slot8 = 255;
      if (pass) {
        if (ref_fast) {
          uint32_t ref0 = (uint32_t)bit_read_at(
              bv.base, bv.rbase_bit + g0.data_bit + (uint64_t)r * g0.W,
              g0.W);
          if (ref0 > g0.count) ref0 = g0.count;
          uint32_t cell = ref0;
          if (ph.n_group_cols > 1) {
            uint32_t ref1 = (uint32_t)bit_read_at(
                bv.base, bv.rbase_bit + g1.data_bit + (uint64_t)r * g1.W,
                g1.W);
            if (ref1 > g1.count) ref1 = g1.count;
            cell += ref1 * dim0;
          }
          uint8_t cs = cell_slot[cell];
          if (cs == 254) {
            int s = g_slot(gtable, build_group_key(bv, ph.n_group_cols,
                                                   gd0, gd1, kl0, kl1, r));
            cs = (s < 0 || s >= 255) ? 255 : (uint8_t)s;
            if (cs == 255) atomicAdd(&counters[1], 1ull);
            cell_slot[cell] = cs;
          }
          slot8 = cs;
        } else {
          int s = g_slot(gtable, build_group_key(bv, ph.n_group_cols, gd0,
                                                 gd1, kl0, kl1, r));
          if (s < 0 || s >= 255) {
            atomicAdd(&counters[1], 1ull);
          } else {
            slot8 = (uint8_t)s;
          }
        }
        if (slot8 != 255) atomicAdd(&cnt[slot8][lane & 3], 1ull);
      }
      if (in) row_slot[row_start + r] = slot8;
    }
  }
  __syncthreads();
  for (uint32_t s = tid; s < OBX_GTABLE_SLOTS; s += WG) {
    unsigned long long c = cnt[s][0] + cnt[s][1] + cnt[s][2] + cnt[s][3];
    if (c) atomicAdd(&gtable[s].count, c);
  }
}

/* ---- aggregate pass: one aggregate (or a fused PROD2+PROD3 pair) ---- */
extern "C" __global__ __launch_bounds__(WG, 2) void k_agg_pass(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_plan_hdr ph, uint32_t a,
    const uint64_t *__restrict__ bitmap, const uint8_t *__restrict__ row_slot,
    gslot *__restrict__ gtable) {
  /* per-slot LDS cells for this pass's aggregate(s): [slot][stripe][2] */
  __shared__ unsigned long long cells[2][OBX_GTABLE_SLOTS][4][2];
  const uint32_t tid = threadIdx.x;
  const uint32_t lane = tid & 63;
  const uint32_t stripe = lane & 3;

  const dev_agg ag = ph.aggs[a];
  const bool fuse_p3 =
      (ag.kind == 4 && a + 1 < ph.n_aggs && ph.aggs[a + 1].kind == 5 &&
       ph.aggs[a + 1].ia == ag.ia && ph.aggs[a + 1].ib == ag.ib);
  const dev_agg ag2 = fuse_p3 ? ph.aggs[a + 1] : ag;
  const bool is_minmax = (ag.kind == 2 || ag.kind == 3);

  for (uint32_t s = tid; s < OBX_GTABLE_SLOTS; s += WG) {
    for (uint32_t f = 0; f < 2; f++) {
      for (uint32_t st = 0; st < 4; st++) {
        cells[f][s][st][0] = (ag.kind == 2) ? (unsigned long long)INT64_MAX
                             : (ag.kind == 3) ? (unsigned long long)INT64_MIN
                                              : 0ull;
        cells[f][s][st][1] = 0;
      }
    }
  }
  __syncthreads();

  unsigned long long *c0 = &cells[0][0][0][0];
  unsigned long long *c1 = &cells[1][0][0][0];

  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const dev_block &cur = blocks[b];
    const uint64_t blk_bit = cur.block_byte * 8;
    blk_view bv;
    bv.base = buf; bv.bit_bias = 0; bv.rbase_bit = blk_bit;
    const uint32_t rows = cur.row_count;
    const uint64_t row_start = dev_block_row_start(&cur);

    col_ctx ca, cb, cc;
    const dev_col *da = nullptr, *db = nullptr, *dc2 = nullptr;
    if (ag.ia != 0xFF) {
      da = &cur.cols[ph.need_cols[ag.ia]];
      ca = make_col_ctx(*da, blk_bit);
    }
    if (ag.ib != 0xFF) {
      db = &cur.cols[ph.need_cols[ag.ib]];
      cb = make_col_ctx(*db, blk_bit);
    }
    uint16_t icx = fuse_p3 ? ag2.ic : ag.ic;
    if (icx != 0xFF && (ag.kind == 5 || fuse_p3)) {
      dc2 = &cur.cols[ph.need_cols[icx]];
      cc = make_col_ctx(*dc2, blk_bit);
    }

    const uint32_t iters = (rows + WG - 1) / WG;
    for (uint32_t it = 0; it < iters; it++) {
      uint32_t r = it * WG + tid;
      if (r >= rows) continue;
      uint8_t slot8 = row_slot[row_start + r];
      if (slot8 == 255) continue;
      uint32_t idx = ((uint32_t)slot8 * 4 + stripe) * 2;
      bool na = false, nb = false, nc = false;
      int64_t va = 0, vb = 0, vc = 0;
      if (ag.ia != 0xFF)
        va = (ca.kind == 4) ? col_value2(bv, cur, *da, r, na)
                            : ctx_value(bv, ca, r, na);
      if (ag.ib != 0xFF)
        vb = (cb.kind == 4) ? col_value2(bv, cur, *db, r, nb)
                            : ctx_value(bv, cb, r, nb);
      if (dc2)
        vc = (cc.kind == 4) ? col_value2(bv, cur, *dc2, r, nc)
                            : ctx_value(bv, cc, r, nc);
      switch (ag.kind) {
        case 0: /* COUNT(col) */
          if (!na) atomicAdd(&c0[idx], 1ull);
          break;
        case 1: /* SUM */
          if (!na) lds_acc_i128(&c0[idx], i128_from_i64(va));
          break;
        case 2: case 3: /* MIN/MAX */
          if (!na) {
            cas_minmax(&c0[idx], va, ag.kind == 2);
            c0[idx + 1] = 1;
          }
          break;
        case 4: /* SUM_PROD2 (+ fused PROD3 mate) */
          if (!na && !nb) {
            i128v p2 = i128_mul_i64(va, ag.one_b - vb);
            lds_acc_i128(&c0[idx], p2);
            if (fuse_p3 && !nc)
              lds_acc_i128(&c1[idx], i128_mul_pos_i64(p2, ag2.one_c + vc));
          }
          break;
        case 5: /* SUM_PROD3 standalone */
          if (!na && !nb && !nc)
            lds_acc_i128(&c0[idx],
                         i128_mul_pos_i64(i128_mul_i64(va, ag.one_b - vb),
                                          ag.one_c + vc));
          break;
        case 6: /* SUM_MUL */
          if (!na && !nb) lds_acc_i128(&c0[idx], i128_mul_i64(va, vb));
          break;
        default: break;
      }
    }
  }
  __syncthreads();

  /* flush: merge the two lane stripes, one global accumulate per slot */
  for (uint32_t s = tid; s < OBX_GTABLE_SLOTS; s += WG) {
    for (uint32_t f = 0; f < (fuse_p3 ? 2u : 1u); f++) {
      uint32_t aa = a + f;
      if (is_minmax) {
        for (uint32_t st = 0; st < 4; st++) {
          if (cells[f][s][st][1]) {
            cas_minmax(&gtable[s].cells[aa][0],
                       (int64_t)cells[f][s][st][0], ag.kind == 2);
            gtable[s].cells[aa][1] = 1;
          }
        }
      } else {
        uint64_t lo = 0, hi = 0;
        for (uint32_t st = 0; st < 4; st++) {
          uint64_t l = cells[f][s][st][0];
          uint64_t nlo = lo + l;
          hi += cells[f][s][st][1] + (nlo < l);
          lo = nlo;
        }
        if (lo | hi) g_acc_i128(gtable[s].cells[aa], lo, hi);
      }
    }
  }
}

/* ---------------- high-cardinality direct-global scan --------------------
 * Growth path past the per-workgroup LDS group table (the reference's
 * ObHashGroupByOp hash table grows unboundedly,
 * ob_exec_hash_struct_vec.h:1718): when the generic kernel reports an LDS
 * table overflow, the host reruns the scan with this kernel. No LDS
 * table: every surviving row probes the global group table (open
 * addressing over OBX_GTABLE_SLOTS) and accumulates with global atomics.
 * Correctness-first slow path — plans that fit the LDS table never take
 * it. AND-combined leaves only (the host keeps OBX_BUF_NOT_ENOUGH for
 * OR-programs). */
__device__ __forceinline__ int g_slot_probe(gslot *gt, uint64_t key) {
  uint32_t idx = (uint32_t)((key * 0x9E3779B97F4A7C15ull) >>
                            (64 - OBX_GTABLE_BIG_SHIFT)) &
                 (OBX_GTABLE_BIG - 1);
  for (int p = 0; p < OBX_GTABLE_BIG; p++) {
    unsigned long long k = gt[idx].key;
    if (k == key) return (int)idx;
    if (k == OBX_KEY_EMPTY) {
      unsigned long long c = atomicCAS(&gt[idx].key, OBX_KEY_EMPTY,
                                       (unsigned long long)key);
      if (c == OBX_KEY_EMPTY || c == key) return (int)idx;
    }
    idx = (idx + 1) & (OBX_GTABLE_BIG - 1);
  }
  return -1;
}

extern "C" __global__ __launch_bounds__(WG, 2) void k_scan_agg_direct(
    const uint8_t *__restrict__ buf, const dev_block *__restrict__ blocks,
    uint32_t n_blocks, const dev_leaf *__restrict__ plan_leaves,
    const blk_leaf *__restrict__ bleaves, const dev_plan_hdr ph,
    gslot *__restrict__ gtable, unsigned long long *__restrict__ counters) {
  const uint32_t tid = threadIdx.x;
  const uint32_t lane = tid & 63, wv = tid >> 6;
  unsigned long long lane_cnt = 0;
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const dev_block &cur = blocks[b];
    const blk_leaf *bl = bleaves + (size_t)b * ph.n_leaves;
    bool skip = false;
    for (uint32_t i = 0; i < ph.n_leaves; i++)
      if (leaf_class(cur, plan_leaves[i], bl[i]) == 0) { skip = true; break; }
    if (skip) continue;
    blk_view bv;
    bv.base = buf;
    bv.bit_bias = 0;
    bv.rbase_bit = 0;
    const dev_col *gd0 = nullptr, *gd1 = nullptr;
    if (ph.n_group_cols > 0) gd0 = &cur.cols[ph.need_cols[ph.group_idx[0]]];
    if (ph.n_group_cols > 1) gd1 = &cur.cols[ph.need_cols[ph.group_idx[1]]];
    const uint32_t kl0 = ph.group_len[0], kl1 = ph.group_len[1];
    const uint32_t rows = cur.row_count;
    for (uint32_t r = tid; r < rows; r += WG) {
      bool live = true;
      for (uint32_t i = 0; live && i < ph.n_leaves; i++)
        live = leaf_match(bv, cur, plan_leaves[i], bl[i], r);
      if (!live) continue;
      lane_cnt++;
      uint64_t key = build_group_key(bv, ph.n_group_cols, gd0, gd1, kl0,
                                     kl1, r);
      int gi = g_slot_probe(gtable, key);
      if (gi < 0) { /* global table full: surfaced as counters[2] */
        atomicAdd(&counters[2], 1ull);
        continue;
      }
      atomicAdd(&gtable[gi].count, 1ull);
      for (uint32_t a = 0; a < ph.n_aggs; a++) {
        const dev_agg ag = ph.aggs[a];
        if (ag.kind == 0 && ag.ia == 0xFF) continue; /* COUNT(*): count */
        bool na = false, nb = false, nc = false;
        int64_t va = 0, vb = 0, vc = 0;
        if (ag.ia != 0xFF)
          va = col_value2(bv, cur, cur.cols[ph.need_cols[ag.ia]], r, na);
        if (ag.kind >= 4 && ag.ib != 0xFF)
          vb = col_value2(bv, cur, cur.cols[ph.need_cols[ag.ib]], r, nb);
        if (ag.kind == 5 && ag.ic != 0xFF)
          vc = col_value2(bv, cur, cur.cols[ph.need_cols[ag.ic]], r, nc);
        switch (ag.kind) {
          case 0: /* COUNT(col) */
            if (!na) g_acc_i128(gtable[gi].cells[a], 1ull, 0ull);
            break;
          case 1: /* SUM */
            if (!na) {
              i128v v = i128_from_i64(va);
              g_acc_i128(gtable[gi].cells[a], v.lo, (uint64_t)v.hi);
            }
            break;
          case 2:
          case 3: /* MIN / MAX */
            if (!na) {
              cas_minmax(&gtable[gi].cells[a][0], va, ag.kind == 2);
              gtable[gi].cells[a][1] = 1;
            }
            break;
          case 4: /* SUM_PROD2 */
            if (!na && !nb) {
              i128v p2 = i128_mul_i64(va, ag.one_b - vb);
              g_acc_i128(gtable[gi].cells[a], p2.lo, (uint64_t)p2.hi);
            }
            break;
          case 5: /* SUM_PROD3 */
            if (!na && !nb && !nc) {
              i128v p3 = i128_mul_pos_i64(i128_mul_i64(va, ag.one_b - vb),
                                          ag.one_c + vc);
              g_acc_i128(gtable[gi].cells[a], p3.lo, (uint64_t)p3.hi);
            }
            break;
          case 6: /* SUM_MUL */
            if (!na && !nb) {
              i128v p = i128_mul_i64(va, vb);
              g_acc_i128(gtable[gi].cells[a], p.lo, (uint64_t)p.hi);
            }
            break;
          default:
            break;
        }
      }
    }
  }
  for (int off = 32; off > 0; off >>= 1)
    lane_cnt += (unsigned long long)__shfl_xor((long long)lane_cnt, off, 64);
  if (lane == 0 && lane_cnt)
    atomicAdd(&counters[8 + ((blockIdx.x * WAVES + wv) & 7)], lane_cnt);
}
