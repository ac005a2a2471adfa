/*
 * kernels.hip — CDNA4 (gfx950) scan -> filter -> aggregate kernels.
 *
 * MI355X-native replacement for the reference's generated per-partition JVM
 * loop (WholeStageCodegen of ColumnTableScan -> Filter ->
 * SnappyHashAggregateExec; ColumnTableScan.scala:636-815,
 * SnappyHashAggregateExec.scala:337-500).  The path is HBM-bandwidth-bound
 * (no dense contraction — MFMA unused by design; DESIGN.md roofline).
 *
 * Structure (third iteration, driven by measurement):
 *  v1 passed per-lane value arrays to helpers -> 112 B/lane HBM-backed
 *     scratch -> 9.6% of HBM peak.
 *  v2 used scalar registers + per-row kind switches -> 1863 basic blocks,
 *     2782 SGPR lane spills (the compiler cannot hoist 8 uniform switches
 *     out of an unrolled row loop) -> 15% of peak.
 *  v3 (this file): per-tile COLUMNAR CONVERSION into LDS, then a branch-free
 *     row phase.
 *     - conversion pass: for each referenced column, one tight coalesced
 *       loop (kind dispatched ONCE per column per chunk) decodes the encoded
 *       body into a canonical LDS image: f64 value (int32/dict index
 *       widened; int64 raw-bitcast to stay exact), plus validity/deleted
 *       bitmap words built with one __ballot per 64 rows.
 *     - row phase: predicates and aggregate factors read the LDS image by
 *       RUNTIME column slot (LDS indexing is free, unlike register arrays),
 *       so there are no per-row branches at all; per-group accumulators stay
 *       in registers (dictionary-direct slots, the reference's
 *       DictionaryOptimizedMapAccessor idea), then one wave-reduced atomic
 *       per value.
 *
 * One launch covers EVERY column batch of the table via a host-built tile
 * array (launch overhead O(1) per query, not O(batches)).
 */
#include <hip/hip_runtime.h>
#include "engine_internal.h"

#define WG 256
#define CHUNK 1024              /* rows converted per LDS round */

__device__ __forceinline__ int bm_get(const uint64_t *bm, int row) {
  return (int)((bm[row >> 6] >> (row & 63)) & 1ull);
}

__device__ __forceinline__ int nonnull_pos(const uint64_t *nullw,
                                           const uint32_t *pfx, int row) {
  uint64_t w = nullw[row >> 6];
  uint64_t mask = (1ull << (row & 63)) - 1ull;
  return row - (int)(pfx[row >> 6] + __popcll(w & mask));
}

__device__ __forceinline__ int patch_find(const int32_t *pos, int n, int row) {
  int lo = 0, hi = n - 1;
  while (lo <= hi) {
    int mid = (lo + hi) >> 1;
    int p = pos[mid];
    if (p == row) return mid;
    if (p < row) lo = mid + 1; else hi = mid - 1;
  }
  return -1;
}

/* general-path read (nulls/patches/any kind); returns 0 if NULL */
__device__ __forceinline__ int read_general(const sn_dev_col &c, int row,
                                            double *vd, long long *vi, int *gid) {
  if (c.patch_n && bm_get(c.patch_bm, row)) {
    int pi = patch_find(c.patch_pos, c.patch_n, row);
    if (pi >= 0) {
      if (c.patch_nullbm && bm_get(c.patch_nullbm, pi)) return 0;
      double pv = c.patch_val[pi];
      if (c.kind == SN_K_DICT16 || c.kind == SN_K_DICT32) {
        *gid = (int)__double2ll_rn(pv);
        return 1;
      }
      if (c.kind == SN_K_F64 || c.kind == SN_K_F32) { *vd = pv; *vi = (long long)pv; }
      else { long long b = __double2ll_rn(pv); *vi = b; *vd = (double)b; }
      return 1;
    }
  }
  int nnp = row;
  if (c.has_nulls) {
    if (bm_get(c.nullw, row)) return 0;
    nnp = nonnull_pos(c.nullw, c.nullpfx, row);
  }
  switch (c.kind) {
    case SN_K_F64: *vd = ((const double *)c.body)[nnp]; *vi = (long long)*vd; break;
    case SN_K_I32: *vi = ((const int32_t *)c.body)[nnp]; *vd = (double)*vi; break;
    case SN_K_I64: *vi = ((const long long *)c.body)[nnp]; *vd = (double)*vi; break;
    case SN_K_F32: *vd = ((const float *)c.body)[nnp]; *vi = (long long)*vd; break;
    case SN_K_I16: *vi = ((const int16_t *)c.body)[nnp]; *vd = (double)*vi; break;
    case SN_K_DICT16: *gid = c.dictmap[(int)(uint16_t)((const int16_t *)c.body)[nnp]]; break;
    case SN_K_DICT32: *gid = c.dictmap[((const int32_t *)c.body)[nnp]]; break;
    case SN_K_BOOLBIT: *vi = bm_get((const uint64_t *)c.body, nnp); *vd = (double)*vi; break;
  }
  return 1;
}

__device__ __forceinline__ double i64_as_f64(long long x) {
  return __longlong_as_double(x);
}
__device__ __forceinline__ long long f64_as_i64(double x) {
  return __double_as_longlong(x);
}

template <int NSLOTS, int NAGGS>
__launch_bounds__(WG, 2)
__global__ void k_scan_agg(sn_dev_plan plan,
                           const sn_dev_batch *__restrict__ batches,
                           const sn_dev_tile *__restrict__ tiles, int ntiles,
                           double *__restrict__ out /* [NSLOTS][2*NAGGS+1] */) {
  const int tid = threadIdx.x;
  const int nused = plan.nused;

  /* dynamic LDS: [nused][CHUNK] f64 values, then [nused][CHUNK/64] validity
   * words, then [CHUNK/64] deleted words */
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double *sval = (double *)smem;
  uint64_t *svalid = (uint64_t *)(smem + (size_t)nused * CHUNK * 8);
  uint64_t *sdead = svalid + (size_t)nused * (CHUNK / 64);

  double sums[NSLOTS][NAGGS];            /* compile-time indexed only */
  double cnts[NSLOTS == 1 ? NAGGS : 1];
  double rc[NSLOTS];
#pragma unroll
  for (int s = 0; s < NSLOTS; s++) {
    rc[s] = 0.0;
#pragma unroll
    for (int a = 0; a < NAGGS; a++) sums[s][a] = 0.0;
  }
  if (NSLOTS == 1) {
#pragma unroll
    for (int a = 0; a < NAGGS; a++) cnts[a] = 0.0;
  }

  const int npreds = plan.npreds, naggs = plan.naggs, ngroup = plan.ngroup;

  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + SN_TILE_ROWS, num_rows);
    const int clean = b.clean;

    for (int base = tile.row_start; base < tile_end; base += CHUNK) {
      const int rows = min(CHUNK, tile_end - base);

      /* ---- conversion pass: one tight loop per referenced column ---- */
      for (int c = 0; c < nused; c++) {
        const sn_dev_col col = b.cols[c];       /* scalar copy (uniform) */
        double *dst = sval + (size_t)c * CHUNK;
        if (clean) {
          switch (col.kind) {
            case SN_K_F64: {
              const double *src = (const double *)col.body + base;
              for (int r = tid; r < rows; r += WG) dst[r] = src[r];
              break;
            }
            case SN_K_I32: {
              const int32_t *src = (const int32_t *)col.body + base;
              for (int r = tid; r < rows; r += WG) dst[r] = (double)src[r];
              break;
            }
            case SN_K_I64: {
              const long long *src = (const long long *)col.body + base;
              for (int r = tid; r < rows; r += WG) dst[r] = i64_as_f64(src[r]);
              break;
            }
            case SN_K_F32: {
              const float *src = (const float *)col.body + base;
              for (int r = tid; r < rows; r += WG) dst[r] = (double)src[r];
              break;
            }
            case SN_K_I16: {
              const int16_t *src = (const int16_t *)col.body + base;
              for (int r = tid; r < rows; r += WG) dst[r] = (double)src[r];
              break;
            }
            case SN_K_DICT16: {
              const int16_t *src = (const int16_t *)col.body + base;
              const int32_t *dm = col.dictmap;
              for (int r = tid; r < rows; r += WG)
                dst[r] = (double)dm[(int)(uint16_t)src[r]];
              break;
            }
            case SN_K_DICT32: {
              const int32_t *src = (const int32_t *)col.body + base;
              const int32_t *dm = col.dictmap;
              for (int r = tid; r < rows; r += WG) dst[r] = (double)dm[src[r]];
              break;
            }
            case SN_K_BOOLBIT: {
              const uint64_t *src = (const uint64_t *)col.body;
              for (int r = tid; r < rows; r += WG)
                dst[r] = (double)bm_get(src, base + r);
              break;
            }
          }
        } else {
          /* general: nulls/patches; validity words via one ballot per 64 rows */
          uint64_t *vw = svalid + (size_t)c * (CHUNK / 64);
          const int is_dict = col.kind == SN_K_DICT16 || col.kind == SN_K_DICT32;
          const int is_i64 = col.kind == SN_K_I64;
          for (int r = tid; r < CHUNK; r += WG) {
            int row = base + r;
            int ok = 0;
            double vd = 0.0; long long vi = 0; int gid = col.null_gid;
            if (row < num_rows)
              ok = read_general(col, row, &vd, &vi, &gid);
            double v = is_dict ? (double)gid : (is_i64 ? i64_as_f64(vi) : vd);
            dst[r] = v;
            uint64_t w = __ballot(ok);
            if ((tid & 63) == 0) vw[r >> 6] = w;
          }
        }
      }
      /* deleted-row words (general only) */
      if (!clean) {
        const uint64_t *del = b.del_bm;
        for (int r = tid; r < CHUNK; r += WG) {
          int row = base + r;
          int dead = (row >= num_rows) || (del && bm_get(del, row));
          uint64_t w = __ballot(dead);
          if ((tid & 63) == 0) sdead[r >> 6] = w;
        }
      }
      __syncthreads();

      /* ---- row phase: branch-free, LDS-indexed by runtime cslot ---- */
      for (int r = tid; r < rows; r += WG) {
        int alive = clean ? 1 : !((sdead[r >> 6] >> (r & 63)) & 1);

#pragma unroll
        for (int i = 0; i < 8; i++) {
          if (i >= npreds) break;
          const sn_dev_pred &p = plan.preds[i];
          const int cs = p.cslot;
          if (!clean)
            alive &= (int)((svalid[(size_t)cs * (CHUNK / 64) + (r >> 6)]
                            >> (r & 63)) & 1ull);
          const double xv = sval[(size_t)cs * CHUNK + r];
          if ((plan.i64_mask >> cs) & 1u) {
            long long x = f64_as_i64(xv);
            if (p.has_lo && (p.lo_strict ? !(x > p.lo_i) : !(x >= p.lo_i))) alive = 0;
            if (p.has_hi && (p.hi_strict ? !(x < p.hi_i) : !(x <= p.hi_i))) alive = 0;
          } else {
            if (p.has_lo && (p.lo_strict ? !(xv > p.lo_d) : !(xv >= p.lo_d))) alive = 0;
            if (p.has_hi && (p.hi_strict ? !(xv < p.hi_d) : !(xv <= p.hi_d))) alive = 0;
          }
        }
        if (__popcll(__ballot(alive)) == 0) continue;

        int slot = 0;
        if (NSLOTS > 1) {
          if (ngroup >= 1)
            slot = (int)sval[(size_t)plan.gcol[0] * CHUNK + r];
          if (ngroup >= 2)
            slot += (int)sval[(size_t)plan.gcol[1] * CHUNK + r];
        }

#pragma unroll
        for (int a = 0; a < NAGGS; a++) {
          if (a >= naggs) break;
          const sn_dev_agg &ag = plan.aggs[a];
          double aval = 1.0;
          int anull = 0;
          if (ag.kind != 1) {
#pragma unroll
            for (int j = 0; j < 3; j++) {
              if (j >= ag.nf) break;
              const int fc = ag.f[j].cslot;
              if (!clean)
                anull |= !(int)((svalid[(size_t)fc * (CHUNK / 64) + (r >> 6)]
                                 >> (r & 63)) & 1ull);
              double x = sval[(size_t)fc * CHUNK + r];
              if ((plan.i64_mask >> fc) & 1u) x = (double)f64_as_i64(x);
              aval = aval * (ag.f[j].add + ag.f[j].mul * x);
            }
          }
          const int m = alive && !anull;
          if (NSLOTS == 1) {
            sums[0][a] += m ? aval : 0.0;
            cnts[a] += m ? 1.0 : 0.0;
          } else {
#pragma unroll
            for (int s = 0; s < NSLOTS; s++)
              sums[s][a] += (m && slot == s) ? aval : 0.0;
          }
        }
        if (NSLOTS == 1) {
          rc[0] += alive ? 1.0 : 0.0;
        } else {
#pragma unroll
          for (int s = 0; s < NSLOTS; s++)
            rc[s] += (alive && slot == s) ? 1.0 : 0.0;
        }
      }
      __syncthreads();   /* LDS reused by the next chunk */
    }
  }

  /* block reduce: wave shuffle reduce, then one global atomic per value per
   * wave (Guideline 12) */
  const int STRIDE = 2 * NAGGS + 1;
#pragma unroll
  for (int s = 0; s < NSLOTS; s++) {
#pragma unroll
    for (int a = 0; a < NAGGS + 1; a++) {
      double x = (a < NAGGS) ? sums[s][a] : rc[s];
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        x += __shfl_down(x, off, 64);
      if ((tid & 63) == 0 && x != 0.0)
        atomicAdd(&out[s * STRIDE + (a < NAGGS ? a : 2 * NAGGS)], x);
    }
  }
  if (NSLOTS == 1) {
#pragma unroll
    for (int a = 0; a < NAGGS; a++) {
      double x = cnts[a];
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        x += __shfl_down(x, off, 64);
      if ((tid & 63) == 0 && x != 0.0)
        atomicAdd(&out[NAGGS + a], x);
    }
  }
}

extern "C" int sn_launch_scan_agg(const sn_dev_plan *plan,
                                  const sn_dev_batch *dev_batches,
                                  const sn_dev_tile *dev_tiles, int32_t ntiles,
                                  double *dev_out, void *stream) {
  hipStream_t s = (hipStream_t)stream;
  int grid = ntiles < 2048 ? (ntiles > 0 ? ntiles : 1) : 2048;
  const int ns = plan->nslots, na = plan->naggs;
  size_t lds = (size_t)plan->nused * CHUNK * 8 +
               (size_t)plan->nused * (CHUNK / 64) * 8 + (CHUNK / 64) * 8 + 16;
  hipError_t err;
#define LAUNCH(S, A)                                                        \
  hipLaunchKernelGGL((k_scan_agg<S, A>), dim3(grid), dim3(WG), lds, s,      \
                     *plan, dev_batches, dev_tiles, ntiles, dev_out);
  if (ns <= 1) {
    if (na <= 2) { LAUNCH(1, 2) }
    else if (na <= 4) { LAUNCH(1, 4) }
    else { LAUNCH(1, 12) }
  } else if (ns <= 4 && na <= 4) { LAUNCH(4, 4) }
  else if (ns <= 8 && na <= 8) { LAUNCH(8, 8) }
  else if (ns <= 16 && na <= 4) { LAUNCH(16, 4) }
  else if (ns <= 16 && na <= 8) { LAUNCH(16, 8) }
  else return (int)hipErrorInvalidValue;
#undef LAUNCH
  err = hipGetLastError();
  return (int)err;
}
