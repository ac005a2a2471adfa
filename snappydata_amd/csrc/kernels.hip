/*
 * kernels.hip — CDNA4 (gfx950) scan -> filter -> aggregate kernels.
 *
 * MI355X-native replacement for the reference's generated per-partition JVM
 * loop (WholeStageCodegen of ColumnTableScan -> Filter ->
 * SnappyHashAggregateExec; ColumnTableScan.scala:636-815,
 * SnappyHashAggregateExec.scala:337-500).  The path is HBM-bandwidth-bound
 * (no dense contraction — MFMA unused by design; see DESIGN.md roofline):
 * the kernels stream encoded column bodies from HBM with coalesced loads,
 * keep per-group accumulators in registers (the reference's
 * DictionaryOptimizedMapAccessor idea: a direct accumulator slot per
 * dictionary combination), and fold block partials into a tiny global
 * accumulator with one atomic per value per wave.
 *
 * One launch covers EVERY column batch of the table: tiles of SN_TILE_ROWS
 * rows are mapped to workgroups through a host-built tile array, so launch
 * overhead is O(1) per query instead of O(batches) (batches are ~24 MB,
 * Literals.scala:129-136 — hundreds per table at SF100).
 *
 * Two row paths, chosen per tile by a wave-uniform branch on the batch
 * descriptor:
 *  - clean: no nulls/deletes/update-patches on any referenced column —
 *    direct fixed-width loads (the bench path).
 *  - general: per-row delete-bitmap check, null bitset + host-built prefix
 *    (nonNullPosition in O(1)), update-patch lookup (binary search over the
 *    host-merged sorted patch list).
 */
#include <hip/hip_runtime.h>
#include "engine_internal.h"

#define WG 256
#define VPT (SN_TILE_ROWS / WG)

__device__ __forceinline__ double sel8(const double v[SN_DEV_MAX_COLS], int i) {
  /* wave-uniform runtime index into a per-lane register array: an unrolled
   * select chain (the index is the same for all lanes; 7 v_cndmask) */
  double r = v[0];
  if (i == 1) r = v[1];
  if (i == 2) r = v[2];
  if (i == 3) r = v[3];
  if (i == 4) r = v[4];
  if (i == 5) r = v[5];
  if (i == 6) r = v[6];
  if (i == 7) r = v[7];
  return r;
}
__device__ __forceinline__ long long sel8i(const long long v[SN_DEV_MAX_COLS], int i) {
  long long r = v[0];
  if (i == 1) r = v[1];
  if (i == 2) r = v[2];
  if (i == 3) r = v[3];
  if (i == 4) r = v[4];
  if (i == 5) r = v[5];
  if (i == 6) r = v[6];
  if (i == 7) r = v[7];
  return r;
}
__device__ __forceinline__ int sel8b(const int v[SN_DEV_MAX_COLS], int i) {
  int r = v[0];
  if (i == 1) r = v[1];
  if (i == 2) r = v[2];
  if (i == 3) r = v[3];
  if (i == 4) r = v[4];
  if (i == 5) r = v[5];
  if (i == 6) r = v[6];
  if (i == 7) r = v[7];
  return r;
}

__device__ __forceinline__ int bm_get(const uint64_t *bm, int row) {
  return (int)((bm[row >> 6] >> (row & 63)) & 1ull);
}

/* nonNullPosition = row - nulls_before(row) via host prefix array */
__device__ __forceinline__ int nonnull_pos(const uint64_t *nullw,
                                           const uint32_t *pfx, int row) {
  uint64_t w = nullw[row >> 6];
  uint64_t mask = (1ull << (row & 63)) - 1ull;
  return row - (int)(pfx[row >> 6] + __popcll(w & mask));
}

/* binary search in sorted patch_pos; returns index or -1 */
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

/* read one column value at `row` on the general path.
 * Returns 0 if NULL (out untouched), else 1 with value in *vd / *vi.
 * For dict group columns, *gid receives the premultiplied global group id. */
__device__ __forceinline__ int read_general(const sn_dev_col &c, int row,
                                            double *vd, long long *vi, int *gid) {
  /* update patch overrides base (UpdatedColumnDecoder semantics) */
  if (c.patch_n && bm_get(c.patch_bm, row)) {
    int pi = patch_find(c.patch_pos, c.patch_n, row);
    if (pi >= 0) {
      if (c.patch_nullbm && bm_get(c.patch_nullbm, pi)) return 0;
      double pv = c.patch_val[pi];
      if (c.kind == SN_K_DICT16 || c.kind == SN_K_DICT32) {
        *gid = (int)(long long)__double2ll_rn(pv);  /* host stored global id */
        return 1;
      }
      if (c.kind == SN_K_F64 || c.kind == SN_K_F32) { *vd = pv; *vi = (long long)pv; }
      else { long long b = (long long)__double2ll_rn(pv); *vi = b; *vd = (double)b; }
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
    case SN_K_DICT16: {
      int idx = (int)(uint16_t)((const int16_t *)c.body)[nnp];
      *gid = c.dictmap[idx];
      break;
    }
    case SN_K_DICT32: {
      int idx = ((const int32_t *)c.body)[nnp];
      *gid = c.dictmap[idx];
      break;
    }
    case SN_K_BOOLBIT: *vi = bm_get((const uint64_t *)c.body, nnp); *vd = (double)*vi; break;
  }
  return 1;
}

template <int NSLOTS, int NAGGS>
__launch_bounds__(WG, 2)
__global__ void k_scan_agg(sn_dev_plan plan,
                           const sn_dev_batch *__restrict__ batches,
                           const sn_dev_tile *__restrict__ tiles, int ntiles,
                           double *__restrict__ out /* [NSLOTS][2*NAGGS+1] */) {
  const int tid = threadIdx.x;
  /* per-lane accumulators: sums per (slot, agg), per-agg non-null counts
   * (keyless only: NSLOTS==1), and row count per slot */
  double sums[NSLOTS][NAGGS];
  double cnts[NSLOTS == 1 ? NAGGS : 1][1];
  double rc[NSLOTS];
#pragma unroll
  for (int s = 0; s < NSLOTS; s++) {
    rc[s] = 0.0;
#pragma unroll
    for (int a = 0; a < NAGGS; a++) sums[s][a] = 0.0;
  }
  if (NSLOTS == 1) {
#pragma unroll
    for (int a = 0; a < NAGGS; a++) cnts[a][0] = 0.0;
  }

  const int npreds = plan.npreds, naggs = plan.naggs, ngroup = plan.ngroup;

  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int row_end = min(tile.row_start + SN_TILE_ROWS, b.num_rows);

    for (int row = tile.row_start + tid; row < row_end; row += WG) {
      double v[SN_DEV_MAX_COLS];
      long long vi[SN_DEV_MAX_COLS];
      int gidv[SN_DEV_MAX_COLS];
      int vnull[SN_DEV_MAX_COLS];
      int alive = 1;

      if (b.clean) {
#pragma unroll
        for (int c = 0; c < SN_DEV_MAX_COLS; c++) {
          const sn_dev_col &col = b.cols[c];
          if (col.body == nullptr) continue;   /* unused slot (uniform) */
          vnull[c] = 0;
          switch (col.kind) {
            case SN_K_F64: v[c] = ((const double *)col.body)[row]; vi[c] = (long long)v[c]; break;
            case SN_K_I32: vi[c] = ((const int32_t *)col.body)[row]; v[c] = (double)vi[c]; break;
            case SN_K_I64: vi[c] = ((const long long *)col.body)[row]; v[c] = (double)vi[c]; break;
            case SN_K_F32: v[c] = ((const float *)col.body)[row]; vi[c] = (long long)v[c]; break;
            case SN_K_I16: vi[c] = ((const int16_t *)col.body)[row]; v[c] = (double)vi[c]; break;
            case SN_K_DICT16:
              gidv[c] = col.dictmap[(int)(uint16_t)((const int16_t *)col.body)[row]];
              break;
            case SN_K_DICT32:
              gidv[c] = col.dictmap[((const int32_t *)col.body)[row]];
              break;
            case SN_K_BOOLBIT: vi[c] = bm_get((const uint64_t *)col.body, row); v[c] = (double)vi[c]; break;
          }
        }
      } else {
        if (b.del_bm && bm_get(b.del_bm, row)) continue;
#pragma unroll
        for (int c = 0; c < SN_DEV_MAX_COLS; c++) {
          const sn_dev_col &col = b.cols[c];
          if (col.body == nullptr) continue;
          gidv[c] = col.null_gid;
          vnull[c] = !read_general(col, row, &v[c], &vi[c], &gidv[c]);
        }
      }

      /* predicates (NULL compares false) */
      for (int i = 0; i < npreds; i++) {
        const sn_dev_pred &p = plan.preds[i];
        int nl = b.clean ? 0 : sel8b(vnull, p.cslot);
        if (nl) { alive = 0; break; }
        if (p.is_double) {
          double x = sel8(v, p.cslot);
          if (p.has_lo && (p.lo_strict ? !(x > p.lo_d) : !(x >= p.lo_d))) { alive = 0; break; }
          if (p.has_hi && (p.hi_strict ? !(x < p.hi_d) : !(x <= p.hi_d))) { alive = 0; break; }
        } else {
          long long x = sel8i(vi, p.cslot);
          if (p.has_lo && (p.lo_strict ? !(x > p.lo_i) : !(x >= p.lo_i))) { alive = 0; break; }
          if (p.has_hi && (p.hi_strict ? !(x < p.hi_i) : !(x <= p.hi_i))) { alive = 0; break; }
        }
      }
      if (__popcll(__ballot(alive)) == 0) continue;   /* whole wave filtered */

      /* group slot */
      int slot = 0;
      if (NSLOTS > 1) {
        if (ngroup >= 1) slot = sel8b(gidv, plan.gcol[0]);
        if (ngroup >= 2) slot += sel8b(gidv, plan.gcol[1]);
      }

      /* aggregate input values (NAGGS compile-time unrolled) */
      double aval[NAGGS];
      int anull[NAGGS];
#pragma unroll
      for (int a = 0; a < NAGGS; a++) {
        aval[a] = 1.0; anull[a] = 0;
        if (a >= naggs) continue;
        const sn_dev_agg &ag = plan.aggs[a];
        if (ag.kind == 1 /* COUNT_STAR */) { aval[a] = 1.0; continue; }
#pragma unroll
        for (int j = 0; j < 3; j++) {
          if (j >= ag.nf) continue;
          const sn_dev_factor &f = ag.f[j];
          if (!b.clean && sel8b(vnull, f.cslot)) anull[a] = 1;
          aval[a] *= f.add + f.mul * sel8(v, f.cslot);
        }
      }

      /* accumulate (slot-predicated register accumulators) */
      if (NSLOTS == 1) {
#pragma unroll
        for (int a = 0; a < NAGGS; a++) {
          int m = alive && !anull[a];
          sums[0][a] += m ? aval[a] : 0.0;
          cnts[a][0] += m ? 1.0 : 0.0;
        }
        rc[0] += alive ? 1.0 : 0.0;
      } else {
#pragma unroll
        for (int s = 0; s < NSLOTS; s++) {
          int m = alive && (slot == s);
          rc[s] += m ? 1.0 : 0.0;
#pragma unroll
          for (int a = 0; a < NAGGS; a++)
            sums[s][a] += m ? aval[a] : 0.0;
        }
      }
    }
  }

  /* block reduce: wave-level shuffle reduce, then one global atomic per
   * value per wave (Guideline 12: partial reduction first, then atomics) */
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
      double x = cnts[a][0];
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
  hipError_t err;
#define LAUNCH(S, A)                                                        \
  hipLaunchKernelGGL((k_scan_agg<S, A>), dim3(grid), dim3(WG), 0, s,        \
                     *plan, dev_batches, dev_tiles, ntiles, dev_out);
  if (ns <= 1) {
    if (na <= 2) { LAUNCH(1, 2) }
    else if (na <= 4) { LAUNCH(1, 4) }
    else { LAUNCH(1, 12) }
  } else if (ns <= 4 && na <= 4) { LAUNCH(4, 4) }
  else if (ns <= 8 && na <= 8) { LAUNCH(8, 8) }
  else if (ns <= 16 && na <= 4) { LAUNCH(16, 4) }
  else if (ns <= 16 && na <= 8) { LAUNCH(16, 8) }
  else return (int)hipErrorInvalidValue;   /* host validates before launch */
#undef LAUNCH
  err = hipGetLastError();
  return (int)err;
}
