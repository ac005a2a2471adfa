/*
 * kernels.hip — CDNA4 (gfx950) scan -> filter -> aggregate kernels (v4).
 *
 * MI355X-native replacement for the reference's generated per-partition JVM
 * loop (WholeStageCodegen of ColumnTableScan -> Filter ->
 * SnappyHashAggregateExec; ColumnTableScan.scala:636-815,
 * SnappyHashAggregateExec.scala:337-500).  HBM-bandwidth-bound by nature
 * (no dense contraction — MFMA unused by design; DESIGN.md roofline).
 *
 * Iteration history (measured on MI355X):
 *  v1 per-lane value arrays -> 112 B/lane scratch -> 9.6% of HBM peak.
 *  v2 scalar registers + per-row kind switches -> 1863 blocks, SGPR spill
 *     storm -> 15%.
 *  v3 columnar LDS conversion + branch-free row phase -> 23% (Q6); grouped
 *     8x8 register accumulators spilled 372 B/lane -> 3.9% (Q1).
 *  v4 (this file):
 *     - conversion loops compile-time unrolled and vectorized (double2 /
 *       int2 loads, ds_write_b128) on full chunks;
 *     - separate grouped kernel that evaluates predicates once into an LDS
 *       (slot, alive) image, then processes aggregates in groups of 4 with
 *       per-lane sums[NSLOTS][4] registers, wave-reducing each chunk into an
 *       LDS block accumulator — register pressure bounded for any slot
 *       count; one global atomic per accumulator per block at the end.
 *
 * One launch covers EVERY column batch via a host-built tile array.
 */
#include <hip/hip_runtime.h>
#include "engine_internal.h"

#define WG 256
#define CHUNK 1024
#define AGRP 4                  /* aggregates per register pass (grouped) */

typedef double double2_t __attribute__((ext_vector_type(2)));
typedef int int2_t __attribute__((ext_vector_type(2)));

/* Global-address-space pointer cast.  Pointers reaching the kernel through
 * descriptor structs are generic, so plain derefs lower to flat_load_* —
 * and flat loads tick BOTH vmcnt and lgkmcnt, so every LDS wait drains all
 * in-flight global loads (measured: the whole prefetch pipeline collapsed).
 * Casting to address_space(1) lowers them to global_load_* (vmcnt only). */
#define GAS __attribute__((address_space(1)))
template <typename T>
__device__ __forceinline__ const GAS T *as_global(const T *p) {
  return (const GAS T *)(unsigned long long)(uintptr_t)p;
}

__device__ __forceinline__ int bm_get(const uint64_t *bm, int row) {
  return (int)((as_global(bm)[row >> 6] >> (row & 63)) & 1ull);
}

__device__ __forceinline__ int nonnull_pos(const uint64_t *nullw,
                                           const uint32_t *pfx, int row) {
  uint64_t w = as_global(nullw)[row >> 6];
  uint64_t mask = (1ull << (row & 63)) - 1ull;
  return row - (int)(as_global(pfx)[row >> 6] + __popcll(w & mask));
}

__device__ __forceinline__ int patch_find(const int32_t *pos_, int n, int row) {
  const GAS int32_t *pos = as_global(pos_);
  int lo = 0, hi = n - 1;
  while (lo <= hi) {
    int mid = (lo + hi) >> 1;
    int p = pos[mid];
    if (p == row) return mid;
    if (p < row) lo = mid + 1; else hi = mid - 1;
  }
  return -1;
}

/* RLE: value of nonNullPosition nnp = vals[first run whose end > nnp] */
__device__ __forceinline__ double rle_value(const sn_dev_col &c, int nnp) {
  const GAS int32_t *ends = (const GAS int32_t *)(uintptr_t)c.rle_ends;
  const GAS double *vals = (const GAS double *)(uintptr_t)c.rle_vals;
  int lo = 0, hi = c.rle_n - 1;
  while (lo < hi) {
    int mid = (lo + hi) >> 1;
    if (ends[mid] > nnp) hi = mid; else lo = mid + 1;
  }
  return vals[lo];
}

/* general-path read (nulls/patches/any kind); returns 0 if NULL */
__device__ __forceinline__ int read_general(const sn_dev_col &c, int row,
                                            double *vd, long long *vi, int *gid) {
  if (c.patch_n && bm_get(c.patch_bm, row)) {
    int pi = patch_find(c.patch_pos, c.patch_n, row);
    if (pi >= 0) {
      if (c.patch_nullbm && bm_get(c.patch_nullbm, pi)) return 0;
      double pv = as_global(c.patch_val)[pi];
      if (c.kind == SN_K_DICT16 || c.kind == SN_K_DICT32) {
        *gid = (int)__double2ll_rn(pv);
        return 1;
      }
      if (c.kind == SN_K_F64 || c.kind == SN_K_F32) { *vd = pv; *vi = (long long)pv; }
      else if (c.kind == SN_K_I64 || c.kind == SN_K_RLE_I64) {
        /* INT64 patch values travel as RAW BITS (decode_delta keeps them
         * exact beyond 2^53); vd keeps the raw-bit double so the LDS image
         * stays in the i64 raw-bit convention */
        *vi = __double_as_longlong(pv); *vd = pv;
      }
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
    case SN_K_F64: *vd = as_global((const double *)c.body)[nnp]; *vi = (long long)*vd; break;
    case SN_K_I32: *vi = as_global((const int32_t *)c.body)[nnp]; *vd = (double)*vi; break;
    case SN_K_I64: *vi = as_global((const long long *)c.body)[nnp]; *vd = (double)*vi; break;
    case SN_K_F32: *vd = as_global((const float *)c.body)[nnp]; *vi = (long long)*vd; break;
    case SN_K_I16: *vi = as_global((const int16_t *)c.body)[nnp]; *vd = (double)*vi; break;
    case SN_K_DICT16: *gid = as_global(c.dictmap)[(int)(uint16_t)as_global((const int16_t *)c.body)[nnp]]; break;
    case SN_K_DICT32: *gid = as_global(c.dictmap)[as_global((const int32_t *)c.body)[nnp]]; break;
    case SN_K_BOOLBIT: *vi = bm_get((const uint64_t *)c.body, nnp); *vd = (double)*vi; break;
    case SN_K_U8: *vi = as_global((const uint8_t *)c.body)[nnp] == 1; *vd = (double)*vi; break;
    case SN_K_S8: *vi = as_global((const int8_t *)c.body)[nnp]; *vd = (double)*vi; break;
    case SN_K_RLE:
      *vd = rle_value(c, nnp);
      *vi = __double2ll_rn(*vd);        /* numeric doubles (i16/i32 runs) */
      break;
    case SN_K_RLE_I64:
      *vd = rle_value(c, nnp);          /* raw i64 bits as double */
      *vi = __double_as_longlong(*vd);  /* bitcast back to the exact value */
      break;
  }
  return 1;
}

/* ---- conversion pass: decode one chunk of every referenced column into the
 * canonical LDS image (f64 values; int64 raw-bitcast; dict -> premultiplied
 * group id as f64).  Vectorized (16 B/lane) on full chunks. ---- */
__device__ __forceinline__ void convert_chunk(
    const sn_dev_batch &b, int nused, int base, int rows, int num_rows,
    double *sval, uint64_t *svalid, uint64_t *sdead) {
  const int tid = threadIdx.x;
  const int clean = b.clean;
  for (int c = 0; c < nused; c++) {
    const sn_dev_col col = b.cols[c];
    double *dst = sval + (size_t)c * CHUNK;
    if (clean) {
      switch (col.kind) {
        case SN_K_F64: {
          const GAS double *src = as_global((const double *)col.body) + base;
          if (rows == CHUNK) {
#pragma unroll
            for (int k = 0; k < CHUNK / (2 * WG); k++) {
              int h = tid + k * WG;   /* pair index */
              ((double2_t *)dst)[h] = ((const GAS double2_t *)src)[h];
            }
          } else {
#pragma unroll
            for (int k = 0; k < CHUNK / WG; k++) {
              int r = tid + k * WG;
              if (r < rows) dst[r] = src[r];
            }
          }
          break;
        }
        case SN_K_I32: {
          const GAS int32_t *src = as_global((const int32_t *)col.body) + base;
          if (rows == CHUNK) {
#pragma unroll
            for (int k = 0; k < CHUNK / (2 * WG); k++) {
              int h = tid + k * WG;
              int2_t x = ((const GAS int2_t *)src)[h];
              double2_t y; y.x = (double)x.x; y.y = (double)x.y;
              ((double2_t *)dst)[h] = y;
            }
          } else {
#pragma unroll
            for (int k = 0; k < CHUNK / WG; k++) {
              int r = tid + k * WG;
              if (r < rows) dst[r] = (double)src[r];
            }
          }
          break;
        }
        case SN_K_I64: {
          const GAS long long *src = as_global((const long long *)col.body) + base;
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = __longlong_as_double(src[r]);
          }
          break;
        }
        case SN_K_F32: {
          const GAS float *src = as_global((const float *)col.body) + base;
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = (double)src[r];
          }
          break;
        }
        case SN_K_I16: {
          const GAS int16_t *src = as_global((const int16_t *)col.body) + base;
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = (double)src[r];
          }
          break;
        }
        case SN_K_DICT16: {
          const GAS int16_t *src = as_global((const int16_t *)col.body) + base;
          const GAS int32_t *dm = as_global(col.dictmap);
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = (double)dm[(int)(uint16_t)src[r]];
          }
          break;
        }
        case SN_K_DICT32: {
          const GAS int32_t *src = as_global((const int32_t *)col.body) + base;
          const GAS int32_t *dm = as_global(col.dictmap);
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = (double)dm[src[r]];
          }
          break;
        }
        case SN_K_BOOLBIT: {
          const uint64_t *src = (const uint64_t *)col.body;  /* via bm_get */
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = (double)bm_get(src, base + r);
          }
          break;
        }
        case SN_K_U8: {
          const GAS uint8_t *src = as_global((const uint8_t *)col.body) + base;
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = (double)(src[r] == 1);
          }
          break;
        }
        case SN_K_S8: {
          const GAS int8_t *src = as_global((const int8_t *)col.body) + base;
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = (double)src[r];
          }
          break;
        }
        case SN_K_RLE:
        case SN_K_RLE_I64: {
          /* RLE_I64 run values are raw i64 bits as doubles — exactly the
           * i64 LDS-image convention, so the same copy is correct */
#pragma unroll
          for (int k = 0; k < CHUNK / WG; k++) {
            int r = tid + k * WG;
            if (r < rows) dst[r] = rle_value(col, base + r);
          }
          break;
        }
      }
    } else {
      uint64_t *vw = svalid + (size_t)c * (CHUNK / 64);
      const int is_dict = col.kind == SN_K_DICT16 || col.kind == SN_K_DICT32;
      const int is_i64 = col.kind == SN_K_I64 || col.kind == SN_K_RLE_I64;
#pragma unroll
      for (int k = 0; k < CHUNK / WG; k++) {
        int r = tid + k * WG;
        int row = base + r;
        int ok = 0;
        double vd = 0.0; long long vi = 0; int gid = col.null_gid;
        if (row < num_rows) ok = read_general(col, row, &vd, &vi, &gid);
        dst[r] = is_dict ? (double)gid : (is_i64 ? __longlong_as_double(vi) : vd);
        uint64_t w = __ballot(ok);
        if ((tid & 63) == 0) vw[r >> 6] = w;
      }
    }
  }
  if (!clean) {
    const uint64_t *del = b.del_bm;
#pragma unroll
    for (int k = 0; k < CHUNK / WG; k++) {
      int r = tid + k * WG;
      int row = base + r;
      int dead = (row >= num_rows) || (del && bm_get(del, row));
      uint64_t w = __ballot(dead);
      if ((tid & 63) == 0) sdead[r >> 6] = w;
    }
  }
  /* short chunks MUST zero the image tail: sval beyond `rows` is raw LDS
   * left by whatever kernel ran last on this CU.  Dead rows never
   * contribute through selects, but the slot-predicated fma accumulators
   * compute fma(0, va, sum) — and fma(0, NaN, s) = NaN, so an ord-ident
   * bit pattern (0xFFF8...) parked in LDS silently poisons every sum in
   * the wave (found by the randomized parity fuzzer as a
   * prior-query-dependent NaN). */
  if (rows < CHUNK) {
    for (int c = 0; c < nused; c++) {
      double *dst = sval + (size_t)c * CHUNK;
      for (int r = rows + tid; r < CHUNK; r += WG) dst[r] = 0.0;
    }
  }
}

/* ---- staged (software-pipelined) conversion for clean full chunks ----
 * T14 async-STAGE split: the NEXT chunk's global loads are issued into
 * registers right after the barrier that frees LDS, so they stay in flight
 * underneath the current chunk's row phase (the single biggest lever here:
 * SQ_WAIT_ANY was 75% of wave cycles with the unstaged conversion).
 * Raw loads are width-based (8/4/2 B); kind conversion happens at the LDS
 * write.  I64 shares the w8 path (its LDS image is the raw bits). */
/* per-tile register copy of the hot column-descriptor fields: reading
 * b.cols[c].body/kind per CHUNK emits serialized global_load_dword +
 * vmcnt(0) chains (measured: a fixed ~0.35 ms per query = the whole Q6
 * budget).  Copy once per tile into compile-time-indexed registers. */
struct ColRegs {
  const void *body;
  const int32_t *dictmap;
  int kind;
};

template <int NC>
__device__ __forceinline__ void hoist_cols(const sn_dev_batch &b, int nused,
                                           ColRegs (&cr)[NC]) {
#pragma unroll
  for (int c = 0; c < NC; c++) {
    cr[c].body = c < nused ? b.cols[c].body : nullptr;
    cr[c].dictmap = c < nused ? b.cols[c].dictmap : nullptr;
    cr[c].kind = c < nused ? b.cols[c].kind : -1;
  }
}

/* one shared register buffer for every width class (64 VGPRs total): raw
 * bits packed into double2 lanes; row mapping is pair-based for every width
 * (pair h = tid + p*WG covers rows 2h, 2h+1) */
template <int NC>
struct Stage {
  double2_t buf[NC][CHUNK / (2 * WG)];
};
__device__ __forceinline__ double pack_u64(unsigned lo, unsigned hi) {
  return __longlong_as_double(((unsigned long long)hi << 32) | lo);
}

__device__ __forceinline__ int col_width_class(int kind) {
  switch (kind) {
    case SN_K_F64: case SN_K_I64: return 8;
    case SN_K_I32: case SN_K_F32: case SN_K_DICT32: return 4;
    case SN_K_I16: case SN_K_DICT16: return 2;
    default: return 0;   /* BOOLBIT: not stageable */
  }
}

template <int NC, int NB>
__device__ __forceinline__ void stage_load(const ColRegs (&cr)[NC], int nused,
                                           int base, Stage<NB> &st) {
  static_assert(NB >= NC || NB == 1, "stage buffer too small");
  const int tid = threadIdx.x;
#pragma unroll
  for (int c = 0; c < NC; c++) {
    if (c >= nused) break;
    const ColRegs &col = cr[c];
    const int w = col_width_class(col.kind);
    if (w == 8) {
      const GAS double2_t *s2 = (const GAS double2_t *)
          as_global((const char *)col.body + (size_t)base * 8);
#pragma unroll
      for (int p = 0; p < CHUNK / (2 * WG); p++) st.buf[NB == 1 ? 0 : c][p] = s2[tid + p * WG];
    } else if (w == 4) {
      /* two int2 pair-loads -> raw bits in buf[c][0].x / .y */
      const GAS int2_t *s2 = (const GAS int2_t *)
          as_global((const char *)col.body + (size_t)base * 4);
      int2_t a0 = s2[tid], a1 = s2[tid + WG];
      st.buf[NB == 1 ? 0 : c][0].x = *(double *)&a0;
      st.buf[NB == 1 ? 0 : c][0].y = *(double *)&a1;
    } else {
      /* two short2 (4 B) pair-loads -> packed into buf[c][0].x */
      const GAS unsigned *s2 = (const GAS unsigned *)
          as_global((const char *)col.body + (size_t)base * 2);
      unsigned a0 = s2[tid], a1 = s2[tid + WG];
      st.buf[NB == 1 ? 0 : c][0].x = pack_u64(a0, a1);
    }
  }
}

template <int NC, int NB>
__device__ __forceinline__ void stage_write(const ColRegs (&cr)[NC], int nused,
                                            Stage<NB> &st, double *sval) {
  static_assert(NB >= NC || NB == 1, "stage buffer too small");
  const int tid = threadIdx.x;
#pragma unroll
  for (int c = 0; c < NC; c++) {
    if (c >= nused) break;
    const ColRegs &col = cr[c];
    double *dst = sval + (size_t)c * CHUNK;
    switch (col.kind) {
      case SN_K_F64: case SN_K_I64:
#pragma unroll
        for (int p = 0; p < CHUNK / (2 * WG); p++)
          ((double2_t *)dst)[tid + p * WG] = st.buf[NB == 1 ? 0 : c][p];
        break;
      case SN_K_I32: case SN_K_F32: case SN_K_DICT32: {
        const GAS int32_t *dm = as_global(col.dictmap);
#pragma unroll
        for (int p = 0; p < CHUNK / (2 * WG); p++) {
          double raw = p == 0 ? st.buf[NB == 1 ? 0 : c][0].x : st.buf[NB == 1 ? 0 : c][0].y;
          int2_t a = *(int2_t *)&raw;
          double2_t y;
          if (col.kind == SN_K_I32) { y.x = (double)a.x; y.y = (double)a.y; }
          else if (col.kind == SN_K_F32) {
            float2 f = *(float2 *)&raw;
            y.x = (double)f.x; y.y = (double)f.y;
          } else { y.x = (double)dm[a.x]; y.y = (double)dm[a.y]; }
          ((double2_t *)dst)[tid + p * WG] = y;
        }
        break;
      }
      case SN_K_I16: case SN_K_DICT16: {
        const GAS int32_t *dm = as_global(col.dictmap);
        unsigned long long raw = __double_as_longlong(st.buf[NB == 1 ? 0 : c][0].x);
#pragma unroll
        for (int p = 0; p < CHUNK / (2 * WG); p++) {
          unsigned half = (unsigned)(raw >> (32 * p));
          int16_t e0 = (int16_t)(half & 0xffff), e1 = (int16_t)(half >> 16);
          double2_t y;
          if (col.kind == SN_K_I16) { y.x = (double)e0; y.y = (double)e1; }
          else {
            y.x = (double)dm[(int)(uint16_t)e0];
            y.y = (double)dm[(int)(uint16_t)e1];
          }
          ((double2_t *)dst)[tid + p * WG] = y;
        }
        break;
      }
      default: break;
    }
  }
}

/* can this batch use the staged pipeline? (clean + every col stageable) */
template <int NC>
__device__ __forceinline__ int batch_stageable(int clean, int nused,
                                               const ColRegs (&cr)[NC]) {
  if (!clean) return 0;
  int ok = 1;
#pragma unroll
  for (int c = 0; c < NC; c++) {
    if (c >= nused) break;
    ok &= col_width_class(cr[c].kind) != 0;
  }
  return ok;
}

/* predicate evaluation for row r (branchless; plan mirrored in LDS) */
__device__ __forceinline__ int eval_preds(const sn_dev_plan *P, int npd, int npi,
                                          int clean, const double *sval,
                                          const uint64_t *svalid,
                                          const uint64_t *sdead, int r) {
  int alive = clean ? 1 : !((sdead[r >> 6] >> (r & 63)) & 1ull);
#pragma unroll
  for (int i = 0; i < 8; i++) {
    if (i >= npd) break;
    const int cs = P->preds_d[i].cslot;
    const double x = sval[(size_t)cs * CHUNK + r];
    alive &= (x >= P->preds_d[i].lo) & (x <= P->preds_d[i].hi);
    if (!clean)
      alive &= (int)((svalid[(size_t)cs * (CHUNK / 64) + (r >> 6)] >> (r & 63)) & 1ull);
  }
#pragma unroll
  for (int i = 0; i < 4; i++) {
    if (i >= npi) break;
    const int cs = P->preds_i[i].cslot;
    const long long x = __double_as_longlong(sval[(size_t)cs * CHUNK + r]);
    alive &= (x >= P->preds_i[i].lo) & (x <= P->preds_i[i].hi);
    if (!clean)
      alive &= (int)((svalid[(size_t)cs * (CHUNK / 64) + (r >> 6)] >> (r & 63)) & 1ull);
  }
  return alive;
}

/* aggregate-factor read: i64 column slots hold RAW BITS in the LDS image
 * (so integer predicates stay exact); aggregate arithmetic is double, so
 * convert on read.  The mask test is scalar-uniform — a scalar branch,
 * free for plans without i64 factors. */
__device__ __forceinline__ double sv_agg(const sn_dev_plan *P,
                                         const double *sval, int cs, int r) {
  const double x = sval[(size_t)cs * CHUNK + r];
  const double xc = (double)__double_as_longlong(x);   /* select, not branch */
  return ((P->i64_mask >> cs) & 1u) ? xc : x;
}

/* aggregate input value: three neutral-padded fmas */
__device__ __forceinline__ double eval_agg(const sn_dev_plan *P,
                                           const sn_dev_agg &A,
                                           const double *sval, int r) {
  return (A.a0 + A.m0 * sv_agg(P, sval, A.c0, r)) *
         (A.a1 + A.m1 * sv_agg(P, sval, A.c1, r)) *
         (A.a2 + A.m2 * sv_agg(P, sval, A.c2, r));
}

/* order-preserving double<->u64 for MIN/MAX accumulators (device twin of
 * sn_f64_ord_h): integer atomicMin/Max implement f64 min/max */
__device__ __forceinline__ unsigned long long f64_ord(double x) {
  unsigned long long b = (unsigned long long)__double_as_longlong(x);
  return (b & 0x8000000000000000ull) ? ~b : (b | 0x8000000000000000ull);
}
#define ORD_MIN_IDENT 0xFFF8000000000000ull   /* enc(+inf) < this; safe top */
#define ORD_MAX_IDENT 0x0000000000000000ull   /* below enc(-inf) */

/* op-aware accumulate into an accumulator cell (LDS or global).
 * Sum cells hold plain doubles; min/max cells hold the ord-u64 encoding
 * bit-cast into the double slot. */
template <typename PTR>
__device__ __forceinline__ void acc_cell(PTR *cell, int op, double va) {
  if (op == 0) {
    (void)atomicAdd((double *)cell, va);
  } else if (op == 1) {
    (void)atomicMin((unsigned long long *)cell, f64_ord(va));
  } else {
    (void)atomicMax((unsigned long long *)cell, f64_ord(va));
  }
}

/* identity value (as raw double bits) for an accumulator cell */
__device__ __forceinline__ double acc_ident(int op) {
  if (op == 0) return 0.0;
  return __longlong_as_double(
      (long long)(op == 1 ? ORD_MIN_IDENT : ORD_MAX_IDENT));
}

/* per-agg factor validity on general batches: 1 when every referenced
 * factor column is non-null for row r (Spark Sum/Average skip null
 * inputs; COUNT(*) has nf == 0 and is always valid) */
__device__ __forceinline__ int agg_valid(const sn_dev_agg &A,
                                         const uint64_t *svalid, int r) {
  int ok = 1;
  if (A.nf >= 1)
    ok &= (int)((svalid[(size_t)A.c0 * (CHUNK / 64) + (r >> 6)] >> (r & 63)) & 1ull);
  if (A.nf >= 2)
    ok &= (int)((svalid[(size_t)A.c1 * (CHUNK / 64) + (r >> 6)] >> (r & 63)) & 1ull);
  if (A.nf >= 3)
    ok &= (int)((svalid[(size_t)A.c2 * (CHUNK / 64) + (r >> 6)] >> (r & 63)) & 1ull);
  return ok;
}

/* wave (64-lane) sum reduction */
__device__ __forceinline__ double wave_sum(double x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_down(x, off, 64);
  return x;
}

/* initialize the alive bitmap words (wave-owned: the wave that processes
 * rows [w*64, w*64+64) is the only writer of word w in every later pass) */
__device__ __forceinline__ void alive_init(uint64_t *salive,
                                           const uint64_t *sdead, int rows,
                                           int clean) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int k = 0; k < CHUNK / WG; k++) {
    const int r = tid + k * WG;
    const int w = r >> 6;
    if ((tid & 63) == 0) {
      uint64_t aw = clean ? ~0ull : ~sdead[w];
      const int rbase = w << 6;
      if (rbase + 64 > rows)
        aw &= rows > rbase ? ((1ull << (rows - rbase)) - 1ull) : 0ull;
      salive[w] = aw;
    }
  }
}

/* fused sweep for <= 3 double predicates (the common fast path): one
 * traversal, params hoisted to ~20 registers, one alive-word RMW */
__device__ __forceinline__ void pred_sweep_fused(const sn_dev_plan *P, int npd,
                                                 int clean, const double *sval,
                                                 const uint64_t *svalid,
                                                 uint64_t *salive) {
  const int tid = threadIdx.x;
  double lo0 = -1e308, hi0 = 1e308, lo1 = lo0, hi1 = hi0, lo2 = lo0, hi2 = hi0;
  int c0 = 0, c1 = 0, c2 = 0;
  if (npd >= 1) { lo0 = P->preds_d[0].lo; hi0 = P->preds_d[0].hi; c0 = P->preds_d[0].cslot; }
  if (npd >= 2) { lo1 = P->preds_d[1].lo; hi1 = P->preds_d[1].hi; c1 = P->preds_d[1].cslot; }
  if (npd >= 3) { lo2 = P->preds_d[2].lo; hi2 = P->preds_d[2].hi; c2 = P->preds_d[2].cslot; }
#pragma unroll 2
  for (int k = 0; k < CHUNK / WG; k++) {
    const int r = tid + k * WG;
    const double x0 = sval[(size_t)c0 * CHUNK + r];
    const double x1 = sval[(size_t)c1 * CHUNK + r];
    const double x2 = sval[(size_t)c2 * CHUNK + r];
    int ok = (x0 >= lo0 && x0 <= hi0) &&
             (npd < 2 || (x1 >= lo1 && x1 <= hi1)) &&
             (npd < 3 || (x2 >= lo2 && x2 <= hi2));
    uint64_t w = __ballot(ok);
    if (!clean) {
      if (npd >= 1) w &= svalid[(size_t)c0 * (CHUNK / 64) + (r >> 6)];
      if (npd >= 2) w &= svalid[(size_t)c1 * (CHUNK / 64) + (r >> 6)];
      if (npd >= 3) w &= svalid[(size_t)c2 * (CHUNK / 64) + (r >> 6)];
    }
    if ((tid & 63) == 0) salive[r >> 6] &= w;
  }
}

/* IN-list membership sweep (bitmap LUT or sorted-list binary search) */
__device__ __forceinline__ void in_sweeps(const sn_dev_plan *P, int clean,
                                          const double *sval,
                                          const uint64_t *svalid,
                                          uint64_t *salive) {
  const int tid = threadIdx.x;
  for (int i = 0; i < P->npreds_in && i < 2; i++) {
    const int cs = P->inp[i].cslot;
    const int is64 = (int)((P->i64_mask >> cs) & 1u);
    const uint64_t *bm = P->inp[i].bm;
    const int64_t *list = P->inp[i].list;
    const long long base = P->inp[i].base;
    const long long lim = (long long)P->inp[i].nwords * 64;
    const int n = P->inp[i].n;
#pragma unroll
    for (int k = 0; k < CHUNK / WG; k++) {
      const int r = tid + k * WG;
      const double x = sval[(size_t)cs * CHUNK + r];
      const long long v = is64 ? __double_as_longlong(x) : (long long)x;
      int ok;
      if (bm) {
        const long long idx = v - base;
        ok = idx >= 0 && idx < lim &&
             (int)((as_global(bm)[idx >> 6] >> (idx & 63)) & 1ull);
      } else {
        const GAS int64_t *ls = as_global(list);
        int lo = 0, hi = n - 1;
        ok = 0;
        while (lo <= hi) {
          const int mid = (lo + hi) >> 1;
          const long long m = (long long)ls[mid];
          if (m == v) { ok = 1; break; }
          if (m < v) lo = mid + 1; else hi = mid - 1;
        }
      }
      uint64_t w = __ballot(ok);
      if (!clean) w &= svalid[(size_t)cs * (CHUNK / 64) + (r >> 6)];
      if ((tid & 63) == 0) salive[r >> 6] &= w;
    }
  }
}

/* one sweep per predicate, params hoisted; wave-owned word updates */
__device__ __forceinline__ void pred_sweeps(const sn_dev_plan *P, int npd,
                                            int npi, int clean,
                                            const double *sval,
                                            const uint64_t *svalid,
                                            uint64_t *salive) {
  const int tid = threadIdx.x;
  if (P->npreds_in > 0) in_sweeps(P, clean, sval, svalid, salive);
  if (npd <= 3 && npi == 0) {
    if (npd > 0) pred_sweep_fused(P, npd, clean, sval, svalid, salive);
    return;
  }
#pragma unroll
  for (int i = 0; i < 8; i++) {
    if (i >= npd) break;
    const int cs = P->preds_d[i].cslot;
    const double lo = P->preds_d[i].lo, hi = P->preds_d[i].hi;
#pragma unroll
    for (int k = 0; k < CHUNK / WG; k++) {
      const int r = tid + k * WG;
      const double x = sval[(size_t)cs * CHUNK + r];
      uint64_t w = __ballot(x >= lo && x <= hi);
      if (!clean) w &= svalid[(size_t)cs * (CHUNK / 64) + (r >> 6)];
      if ((tid & 63) == 0) salive[r >> 6] &= w;
    }
  }
#pragma unroll
  for (int i = 0; i < 4; i++) {
    if (i >= npi) break;
    const int cs = P->preds_i[i].cslot;
    const long long lo = P->preds_i[i].lo, hi = P->preds_i[i].hi;
#pragma unroll
    for (int k = 0; k < CHUNK / WG; k++) {
      const int r = tid + k * WG;
      const long long x = __double_as_longlong(sval[(size_t)cs * CHUNK + r]);
      uint64_t w = __ballot(x >= lo && x <= hi);
      if (!clean) w &= svalid[(size_t)cs * (CHUNK / 64) + (r >> 6)];
      if ((tid & 63) == 0) salive[r >> 6] &= w;
    }
  }
}

/* broadcast-dimension probe sweep (HashJoinExec probe, device-side):
 * open-address linear-probe table in HBM (L2/L3-resident by the 100 MB
 * HashJoinSize design bound); filters alive and, for group-by-dim-attr
 * joins, assigns the group slot from the dimension payload. */
__device__ __forceinline__ unsigned long long mix64(unsigned long long x) {
  x += 0x9E3779B97f4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

__device__ __forceinline__ void probe_sweep(const sn_dev_plan *P,
                                            const double *sval,
                                            uint64_t *salive,
                                            int16_t *sslot) {
  const int tid = threadIdx.x;
  const GAS long long *jk = (const GAS long long *)(uintptr_t)P->jkeys;
  const GAS int32_t *jp = (const GAS int32_t *)(uintptr_t)P->jpayload;
  const int cs = P->jcslot;
  const unsigned mask = (1u << P->jcap_log2) - 1;
  const int is_i64 = (P->i64_mask >> cs) & 1u;
  const GAS int32_t *lut = (const GAS int32_t *)(uintptr_t)P->jlut;
  const long long lmin = P->jlut_min, lmax = P->jlut_max;
#pragma unroll 2
  for (int k = 0; k < CHUNK / WG; k++) {
    const int r = tid + k * WG;
    const double xv = sval[(size_t)cs * CHUNK + r];
    const long long key = is_i64 ? __double_as_longlong(xv) : (long long)xv;
    int pay = -1;
    if (lut) {
      /* dense span: one dependent load, no hash chain */
      if (key >= lmin && key <= lmax) pay = lut[key - lmin];
    } else {
      unsigned h = (unsigned)mix64((unsigned long long)key) & mask;
      while (true) {
        long long k0 = jk[h];
        if (k0 == key) { pay = jp[h]; break; }
        if (k0 == LLONG_MIN) break;
        h = (h + 1) & mask;
      }
    }
    const uint64_t w = __ballot(pay >= 0);
    if ((tid & 63) == 0) salive[r >> 6] &= w;
    if (sslot) sslot[r] = (int16_t)(pay > 0 ? pay : 0);
  }
}

/* ================= keyless kernel ================= */
template <int NAGGS, int NC>
__launch_bounds__(WG, NAGGS <= 4 ? 4 : 2)
__global__ void k_keyless(sn_dev_plan plan,
                          const sn_dev_plan *__restrict__ plan_g,
                          const sn_dev_batch *__restrict__ batches,
                          const sn_dev_tile *__restrict__ tiles, int ntiles,
                          double *__restrict__ out /* [2*NAGGS+1] */) {
  const int tid = threadIdx.x;
  const int nused = plan.nused;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double *sval = (double *)smem;
  uint64_t *svalid = (uint64_t *)(smem + (size_t)nused * CHUNK * 8);
  uint64_t *sdead = svalid + (size_t)nused * (CHUNK / 64);
  uint64_t *salive = sdead + CHUNK / 64;
  /* plan mirrored into LDS: row-phase field reads become broadcast ds_reads
   * instead of SGPR-spilled kernarg loads */
  sn_dev_plan *P = (sn_dev_plan *)(salive + CHUNK / 64 + 2);
  {
    const GAS unsigned *src = (const GAS unsigned *)(uintptr_t)plan_g;
    unsigned *dst = (unsigned *)P;
    for (unsigned i = tid; i < sizeof(sn_dev_plan) / 4; i += WG) dst[i] = src[i];
  }

  double sums[NAGGS], cnts[NAGGS], rcnt = 0.0;
#pragma unroll
  for (int a = 0; a < NAGGS; a++) { sums[a] = 0.0; cnts[a] = 0.0; }
  const int naggs = plan.naggs;
  const int npd = plan.npreds_d, npi = plan.npreds_i;
  constexpr int STAGED = 1;

  Stage<STAGED ? NC : 1> st;
  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + SN_TILE_ROWS, num_rows);
    const int clean = b.clean;
    ColRegs cr[NC];
    hoist_cols(b, nused, cr);
    const int pipe = STAGED && batch_stageable(clean, nused, cr);

    int staged = 0;
    if (pipe && tile.row_start + CHUNK <= tile_end) {
      stage_load(cr, nused, tile.row_start, st);
      staged = 1;
    }
    for (int base = tile.row_start; base < tile_end; base += CHUNK) {
      const int rows = min(CHUNK, tile_end - base);
      if (staged) stage_write(cr, nused, st, sval);
      else convert_chunk(b, nused, base, rows, num_rows, sval, svalid, sdead);
      __syncthreads();
      /* prefetch the NEXT full chunk under this chunk's row phase */
      const int nbase = base + CHUNK;
      const int next_staged = pipe && nbase + CHUNK <= tile_end;
      if (next_staged) stage_load(cr, nused, nbase, st);

      /* fully-fused single pass (clean chunks, <=3 double preds, no join):
       * one traversal computes predicates, aggregates and counts — the
       * minimal LDS-read structure (matches the membw probe's shape) */
      if (clean && npd <= 3 && npi == 0 && !plan.jkeys &&
          plan.npreds_in == 0) {
        double lo0 = -1e308, hi0 = 1e308, lo1 = lo0, hi1 = hi0, lo2 = lo0, hi2 = hi0;
        int c0 = 0, c1 = 0, c2 = 0;
        if (npd >= 1) { lo0 = P->preds_d[0].lo; hi0 = P->preds_d[0].hi; c0 = P->preds_d[0].cslot; }
        if (npd >= 2) { lo1 = P->preds_d[1].lo; hi1 = P->preds_d[1].hi; c1 = P->preds_d[1].cslot; }
        if (npd >= 3) { lo2 = P->preds_d[2].lo; hi2 = P->preds_d[2].hi; c2 = P->preds_d[2].cslot; }
#pragma unroll 2
        for (int k = 0; k < CHUNK / WG; k++) {
          const int r = tid + k * WG;
          const int inr = r < rows;
          const double x0 = sval[(size_t)c0 * CHUNK + r];
          const double x1 = sval[(size_t)c1 * CHUNK + r];
          const double x2 = sval[(size_t)c2 * CHUNK + r];
          const int ok = inr &&
              (npd < 1 || (x0 >= lo0 && x0 <= hi0)) &&
              (npd < 2 || (x1 >= lo1 && x1 <= hi1)) &&
              (npd < 3 || (x2 >= lo2 && x2 <= hi2));
          if (__popcll(__ballot(ok)) == 0) continue;
#pragma unroll
          for (int a = 0; a < NAGGS; a++) {
            if (a >= naggs) break;
            const double va = eval_agg(P, P->aggs[a], sval, r);
            sums[a] += ok ? va : 0.0;
            cnts[a] += ok ? 1.0 : 0.0;
          }
          rcnt += ok ? 1.0 : 0.0;
        }
        __syncthreads();
        staged = next_staged;
        continue;
      }

      /* ---- row phase as sweep passes: each wave owns its 64-row words of
       * the alive bitmap, so pred/agg passes need no barriers; plan params
       * hoisted per pass (row-invariant LDS reads once, not per row) ---- */
      alive_init(salive, sdead, rows, clean);
      pred_sweeps(P, npd, npi, clean, sval, svalid, salive);
      if (plan.jkeys) probe_sweep(P, sval, salive, nullptr);
#pragma unroll
      for (int a = 0; a < NAGGS; a++) {
        if (a >= naggs) break;
        const sn_dev_agg A = P->aggs[a];
#pragma unroll 2
        for (int k = 0; k < CHUNK / WG; k++) {
          const int r = tid + k * WG;
          uint64_t w = salive[r >> 6];
          if (!clean) {
            if (A.nf >= 1) w &= svalid[(size_t)A.c0 * (CHUNK / 64) + (r >> 6)];
            if (A.nf >= 2) w &= svalid[(size_t)A.c1 * (CHUNK / 64) + (r >> 6)];
            if (A.nf >= 3) w &= svalid[(size_t)A.c2 * (CHUNK / 64) + (r >> 6)];
          }
          if (w == 0) continue;
          const int m = (int)((w >> (tid & 63)) & 1ull);
          const double val = eval_agg(P, A, sval, r);
          sums[a] += m ? val : 0.0;
          cnts[a] += m ? 1.0 : 0.0;
        }
      }
#pragma unroll 2
      for (int k = 0; k < CHUNK / WG; k++) {
        const int r = tid + k * WG;
        rcnt += (double)((salive[r >> 6] >> (tid & 63)) & 1ull);
      }
      __syncthreads();
      staged = next_staged;
    }
  }

  /* block partials to scratch (plain stores) — per-wave atomicAdd to the
   * same few global addresses serialized at ~45ns each (8K+ per address =
   * a fixed ~0.35 ms per query); a tiny reduce kernel sums the partials */
  double *bacc = (double *)(P + 1);
  {
    const int NV = 2 * NAGGS + 1;
    for (int i = tid; i < NV; i += WG) bacc[i] = 0.0;
    __syncthreads();
#pragma unroll
    for (int a = 0; a < NAGGS; a++) {
      double x = wave_sum(sums[a]);
      if ((tid & 63) == 0 && x != 0.0) atomicAdd(bacc + a, x);
      x = wave_sum(cnts[a]);
      if ((tid & 63) == 0 && x != 0.0) atomicAdd(bacc + NAGGS + a, x);
    }
    double x = wave_sum(rcnt);
    if ((tid & 63) == 0 && x != 0.0) atomicAdd(bacc + 2 * NAGGS, x);
    __syncthreads();
    for (int i = tid; i < NV; i += WG)
      out[(size_t)blockIdx.x * NV + i] = bacc[i];
  }
}

/* ================= grouped kernel =================
 * Aggregates processed in groups of AGRP with per-lane sums[NSLOTS][AGRP]
 * registers; after each chunk the registers wave-reduce into the LDS block
 * accumulator (nslots x (naggs+1)), which flushes once per block to global
 * atomics.  Requires non-nullable aggregate inputs (engine-validated);
 * output layout [slot][2*NA+1]: sums, counts(=rowcount written host-side),
 * rowcount. */
template <int NSLOTS, int NC>
__launch_bounds__(WG, 2)
__global__ void k_grouped(sn_dev_plan plan,
                          const sn_dev_plan *__restrict__ plan_g,
                          const sn_dev_batch *__restrict__ batches,
                          const sn_dev_tile *__restrict__ tiles, int ntiles,
                          double *__restrict__ out, int out_stride) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int nused = plan.nused;
  const int naggs = plan.naggs, ngroup = plan.ngroup;
  const int npd = plan.npreds_d, npi = plan.npreds_i;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  double *sval = (double *)smem;
  uint64_t *svalid = (uint64_t *)(smem + (size_t)nused * CHUNK * 8);
  uint64_t *sdead = svalid + (size_t)nused * (CHUNK / 64);
  uint64_t *salive = sdead + CHUNK / 64;
  int16_t *sslot = (int16_t *)(salive + CHUNK / 64);
  /* block accumulator: [NSLOTS][naggs+1], init once */
  double *bacc = (double *)(sslot + CHUNK);
  sn_dev_plan *P = (sn_dev_plan *)(bacc + NSLOTS * (naggs + 1) + 2);
  {
    const GAS unsigned *src = (const GAS unsigned *)(uintptr_t)plan_g;
    unsigned *dst = (unsigned *)P;
    for (unsigned i = tid; i < sizeof(sn_dev_plan) / 4; i += WG) dst[i] = src[i];
  }

  for (int i = tid; i < NSLOTS * (naggs + 1); i += WG) bacc[i] = 0.0;
  __syncthreads();

  constexpr int STAGED = 1;
  Stage<STAGED ? NC : 1> st;
  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + SN_TILE_ROWS, num_rows);
    const int clean = b.clean;
    ColRegs cr[NC];
    hoist_cols(b, nused, cr);
    const int pipe = STAGED && batch_stageable(clean, nused, cr);

    int staged = 0;
    if (pipe && tile.row_start + CHUNK <= tile_end) {
      stage_load(cr, nused, tile.row_start, st);
      staged = 1;
    }
    for (int base = tile.row_start; base < tile_end; base += CHUNK) {
      const int rows = min(CHUNK, tile_end - base);
      if (staged) stage_write(cr, nused, st, sval);
      else convert_chunk(b, nused, base, rows, num_rows, sval, svalid, sdead);
      __syncthreads();
      const int nbase = base + CHUNK;
      const int next_staged = pipe && nbase + CHUNK <= tile_end;
      if (next_staged) stage_load(cr, nused, nbase, st);

      /* pass A: alive bitmap via sweeps + slot per row (wave-owned words,
       * no barrier needed before this block's own later passes) */
      alive_init(salive, sdead, rows, clean);
      pred_sweeps(P, npd, npi, clean, sval, svalid, salive);
      if (plan.jkeys) {
        /* join: filter by probe; group-by-dim-attr takes its slot from the
         * dimension payload */
        probe_sweep(P, sval, salive, plan.jmode == 1 ? sslot : nullptr);
      }
      if (!(plan.jkeys && plan.jmode == 1)) {
        const int gc0 = plan.gcol[0], gc1 = plan.gcol[1];
#pragma unroll
        for (int k = 0; k < CHUNK / WG; k++) {
          const int r = tid + k * WG;
          int slot = 0;
          if (ngroup >= 1)
            slot = (int)(((long long)sval[(size_t)gc0 * CHUNK + r] -
                          P->gbase[0]) * P->gmul0);
          if (ngroup >= 2)
            slot += (int)((long long)sval[(size_t)gc1 * CHUNK + r] -
                          P->gbase[1]);
          sslot[r] = (int16_t)slot;
        }
      } else if (ngroup >= 1) {
        /* composite GROUP BY dim_attr, fact_col: probe_sweep staged the
         * attr gid; widen it by the dense fact-slot space (dead rows keep
         * garbage slots — they are never accumulated) */
        const int gc0 = plan.gcol[0];
#pragma unroll
        for (int k = 0; k < CHUNK / WG; k++) {
          const int r = tid + k * WG;
          const int slot = (int)sslot[r] * P->jslot_mul +
              (int)((long long)sval[(size_t)gc0 * CHUNK + r] - P->gbase[0]);
          sslot[r] = (int16_t)slot;
        }
      }

      /* rowcount per slot */
      {
        double rc[NSLOTS];
#pragma unroll
        for (int s = 0; s < NSLOTS; s++) rc[s] = 0.0;
#pragma unroll 2
        for (int k = 0; k < CHUNK / WG; k++) {
          int r = tid + k * WG;
          int alive = (int)((salive[r >> 6] >> (r & 63)) & 1ull);
          int slot = sslot[r];
#pragma unroll
          for (int s = 0; s < NSLOTS; s++)
            rc[s] += (alive && slot == s) ? 1.0 : 0.0;
        }
#pragma unroll
        for (int s = 0; s < NSLOTS; s++) {
          double x = wave_sum(rc[s]);
          if ((tid & 63) == 0 && x != 0.0)
            atomicAdd(&bacc[s * (naggs + 1) + naggs], x);
        }
      }

      /* pass B: aggregate PAIRS per sweep, params hoisted, slot-predicated
       * register accumulators (2 x NSLOTS live at a time) */
      for (int a0 = 0; a0 < naggs; a0 += 2) {
        const sn_dev_agg A = P->aggs[a0];
        const int has2 = a0 + 1 < naggs;
        const sn_dev_agg B2 = P->aggs[has2 ? a0 + 1 : a0];
        double sa[NSLOTS], sb[NSLOTS];
#pragma unroll
        for (int s = 0; s < NSLOTS; s++) { sa[s] = 0.0; sb[s] = 0.0; }
#pragma unroll 4
        for (int k = 0; k < CHUNK / WG; k++) {
          const int r = tid + k * WG;
          const uint64_t w = salive[r >> 6];
          if (w == 0) continue;
          const int m = (int)((w >> (tid & 63)) & 1ull);
          const int slot = sslot[r];
          const double va = eval_agg(P, A, sval, r);
          const double vb = eval_agg(P, B2, sval, r);
#pragma unroll
          for (int s = 0; s < NSLOTS; s++) {
            /* fma(md, x, acc) is bit-identical to the select+add form and
             * halves the VALU ops per slot */
            const double md = (m && slot == s) ? 1.0 : 0.0;
            sa[s] = __builtin_fma(md, va, sa[s]);
            sb[s] = __builtin_fma(md, vb, sb[s]);
          }
        }
#pragma unroll
        for (int s = 0; s < NSLOTS; s++) {
          double x = wave_sum(sa[s]);
          if ((tid & 63) == 0 && x != 0.0)
            atomicAdd(&bacc[s * (naggs + 1) + a0], x);
          if (has2) {
            x = wave_sum(sb[s]);
            if ((tid & 63) == 0 && x != 0.0)
              atomicAdd(&bacc[s * (naggs + 1) + a0 + 1], x);
          }
        }
      }
      __syncthreads();
      staged = next_staged;
    }
  }

  /* flush block accumulator to a per-block scratch row (plain stores);
   * the reduce kernel folds rows into the final [slot][stride] layout.
   * Row width uses the RUNTIME slot count (the engine sizes scratch by it;
   * the template NSLOTS is only the register-capacity ceiling). */
  __syncthreads();
  const int NV = plan.nslots * (naggs + 1);
  for (int i = tid; i < NV; i += WG)
    out[(size_t)blockIdx.x * NV + i] = bacc[i];
  (void)wid; (void)out_stride;
}

/* ---- LDS-accumulator grouped kernel (large slot counts: 17..1024) ----
 * The ByteBufferHashMap/SHAMap analogue for dictionary group keys: the
 * premultiplied global dictionary id IS the table slot (the reference's
 * DictionaryOptimizedMapAccessor direct-array idea scaled up), and the
 * per-block accumulator lives in LDS with f64 atomics — collisions are
 * per-wave lanes hitting one slot, rare at high cardinality.  One LDS
 * accumulator per block accumulates across all its tiles and flushes to a
 * per-block scratch row at the end (grid is capped so scratch stays small).
 * Open-address hashing of arbitrary (non-dictionary) keys is round-2. */
__global__ __launch_bounds__(WG, 1)
void k_grouped_lds(sn_dev_plan plan,
                   const sn_dev_plan *__restrict__ plan_g,
                   const sn_dev_batch *__restrict__ batches,
                   const sn_dev_tile *__restrict__ tiles, int ntiles,
                   double *__restrict__ out, int out_stride) {
  const int tid = threadIdx.x;
  const int nused = plan.nused;
  const int naggs = plan.naggs, ngroup = plan.ngroup;
  const int npd = plan.npreds_d, npi = plan.npreds_i;
  const int nslots = plan.nslots;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  double *sval = (double *)smem;
  uint64_t *svalid = (uint64_t *)(smem + (size_t)nused * CHUNK * 8);
  uint64_t *sdead = svalid + (size_t)nused * (CHUNK / 64);
  uint64_t *salive = sdead + CHUNK / 64;
  int16_t *sslot = (int16_t *)(salive + CHUNK / 64 + 2);   /* join-group */
  sn_dev_plan *P = (sn_dev_plan *)((char *)sslot + CHUNK * 2);
  double *bacc = (double *)(P + 1);   /* [nslots][naggs+1] */
  {
    const GAS unsigned *src = (const GAS unsigned *)(uintptr_t)plan_g;
    unsigned *dst = (unsigned *)P;
    for (unsigned i = tid; i < sizeof(sn_dev_plan) / 4; i += WG) dst[i] = src[i];
  }
  /* pac: accumulator rows widen to [sums][per-agg counts][rowcount];
   * min/max cells initialize to their ord-encoding identity (read ops from
   * the by-value plan — the LDS mirror copy has no barrier yet) */
  const int pac = plan.pac;
  const int na1 = pac ? 2 * naggs + 1 : naggs + 1;
  const int NV = nslots * na1;
  for (int i = tid; i < NV; i += WG) {
    const int a = i % na1;
    bacc[i] = a < naggs ? acc_ident(plan.aggs[a].op) : 0.0;
  }
  __syncthreads();

  const int gc0 = plan.gcol[0], gc1 = plan.gcol[1];
  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + SN_TILE_ROWS, num_rows);
    const int clean = b.clean;

    for (int base = tile.row_start; base < tile_end; base += CHUNK) {
      const int rows = min(CHUNK, tile_end - base);
      convert_chunk(b, nused, base, rows, num_rows, sval, svalid, sdead);
      __syncthreads();
      alive_init(salive, sdead, rows, clean);
      pred_sweeps(P, npd, npi, clean, sval, svalid, salive);
      const int jslot = plan.jkeys && plan.jmode == 1;
      if (plan.jkeys) probe_sweep(P, sval, salive, jslot ? sslot : nullptr);

#pragma unroll 2
      for (int k = 0; k < CHUNK / WG; k++) {
        const int r = tid + k * WG;
        const uint64_t w = salive[r >> 6];
        if (w == 0) continue;
        const int m = (int)((w >> (tid & 63)) & 1ull);
        if (!m) continue;
        int slot = 0;
        if (jslot) {
          slot = sslot[r];
          if (ngroup >= 1)   /* composite dim_attr x fact_col */
            slot = slot * P->jslot_mul +
                   (int)((long long)sval[(size_t)gc0 * CHUNK + r] -
                         P->gbase[0]);
        } else {
          if (ngroup >= 1)
            slot = (int)(((long long)sval[(size_t)gc0 * CHUNK + r] -
                          P->gbase[0]) * P->gmul0);
          if (ngroup >= 2)
            slot += (int)((long long)sval[(size_t)gc1 * CHUNK + r] -
                          P->gbase[1]);
        }
        double *row_acc = bacc + (size_t)slot * na1;
        for (int a = 0; a < naggs; a++) {
          const sn_dev_agg &A = P->aggs[a];
          if (pac) {
            const int av = clean ? 1 : agg_valid(A, svalid, r);
            if (!av) continue;
            atomicAdd(&row_acc[naggs + a], 1.0);
          }
          acc_cell(&row_acc[a], A.op, eval_agg(P, A, sval, r));
        }
        atomicAdd(&row_acc[na1 - 1], 1.0);
      }
      __syncthreads();
    }
  }

  __syncthreads();
  for (int i = tid; i < NV; i += WG)
    out[(size_t)blockIdx.x * NV + i] = bacc[i];
  (void)out_stride;
}

/* ---- register-accumulator grouped kernel (small shapes: NSLOTS x NA
 * accumulators live across the WHOLE kernel) ----
 * The sweep kernel's per-chunk wave reductions made it LDS-latency-bound
 * (measured: 61% of wave cycles parked, LDS array 40% busy, HBM 12%).
 * Here each lane accumulates slot-predicated sums in registers through all
 * tiles and reduces ONCE at kernel end, one LDS read per referenced value
 * per row.  Dict-group plans with nslots <= NSLOTS and deduped aggs <= NA;
 * join-group and larger shapes use k_grouped. */
template <int NSLOTS, int NA, int NC, int STAGED = 1>
__launch_bounds__(WG, STAGED ? 2 : 3)
__global__ void k_grouped_reg(sn_dev_plan plan,
                              const sn_dev_plan *__restrict__ plan_g,
                              const sn_dev_batch *__restrict__ batches,
                              const sn_dev_tile *__restrict__ tiles, int ntiles,
                              double *__restrict__ out, int out_stride) {
  const int tid = threadIdx.x;
  const int nused = plan.nused;
  const int naggs = plan.naggs, ngroup = plan.ngroup;
  const int npd = plan.npreds_d, npi = plan.npreds_i;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  double *sval = (double *)smem;
  uint64_t *svalid = (uint64_t *)(smem + (size_t)nused * CHUNK * 8);
  uint64_t *sdead = svalid + (size_t)nused * (CHUNK / 64);
  uint64_t *salive = sdead + CHUNK / 64;
  sn_dev_plan *P = (sn_dev_plan *)(salive + CHUNK / 64 + 2);
  double *bacc = (double *)(P + 1);
  {
    const GAS unsigned *src = (const GAS unsigned *)(uintptr_t)plan_g;
    unsigned *dst = (unsigned *)P;
    for (unsigned i = tid; i < sizeof(sn_dev_plan) / 4; i += WG) dst[i] = src[i];
  }
  __syncthreads();

  const int gc0 = plan.gcol[0], gc1 = plan.gcol[1];

  double sums[NSLOTS][NA];
  double rc[NSLOTS];
#pragma unroll
  for (int s = 0; s < NSLOTS; s++) {
    rc[s] = 0.0;
#pragma unroll
    for (int a = 0; a < NA; a++) sums[s][a] = 0.0;
  }

  Stage<STAGED ? NC : 1> st;
  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + SN_TILE_ROWS, num_rows);
    const int clean = b.clean;
    ColRegs cr[NC];
    hoist_cols(b, nused, cr);
    const int pipe = STAGED && batch_stageable(clean, nused, cr);

    int staged = 0;
    if (STAGED && pipe && tile.row_start + CHUNK <= tile_end) {
      if constexpr (STAGED) stage_load(cr, nused, tile.row_start, st);
      staged = 1;
    }
    for (int base = tile.row_start; base < tile_end; base += CHUNK) {
      const int rows = min(CHUNK, tile_end - base);
      if (STAGED && staged) { if constexpr (STAGED) stage_write(cr, nused, st, sval); }
      else convert_chunk(b, nused, base, rows, num_rows, sval, svalid, sdead);
      __syncthreads();
      const int nbase = base + CHUNK;
      const int next_staged = pipe && nbase + CHUNK <= tile_end;
      if (STAGED && next_staged) { if constexpr (STAGED) stage_load(cr, nused, nbase, st); }

      const int inline_preds = clean && npi == 0 && !plan.jkeys &&
                               plan.npreds_in == 0;
      if (!inline_preds) {
        alive_init(salive, sdead, rows, clean);
        pred_sweeps(P, npd, npi, clean, sval, svalid, salive);
        if (plan.jkeys) probe_sweep(P, sval, salive, nullptr);
      }

#pragma unroll 2
      for (int k = 0; k < CHUNK / WG; k++) {
        const int r = tid + k * WG;
        int m;
        if (inline_preds) {
          m = r < rows;
          for (int i = 0; i < npd; i++) {
            const double x = sval[(size_t)P->preds_d[i].cslot * CHUNK + r];
            m &= (x >= P->preds_d[i].lo) & (x <= P->preds_d[i].hi);
          }
          if (__popcll(__ballot(m)) == 0) continue;
        } else {
          const uint64_t w = salive[r >> 6];
          if (w == 0) continue;
          m = (int)((w >> (tid & 63)) & 1ull);
        }
        int slot = 0;
        if (ngroup >= 1)
          slot = (int)(((long long)sval[(size_t)gc0 * CHUNK + r] -
                        P->gbase[0]) * P->gmul0);
        if (ngroup >= 2)
          slot += (int)((long long)sval[(size_t)gc1 * CHUNK + r] -
                        P->gbase[1]);
        double va[NA];
#pragma unroll
        for (int a = 0; a < NA; a++) {
          if (a >= naggs) { va[a] = 0.0; continue; }
          va[a] = eval_agg(P, P->aggs[a], sval, r);
        }
#pragma unroll
        for (int s = 0; s < NSLOTS; s++) {
          const double md = (m && slot == s) ? 1.0 : 0.0;
          rc[s] += md;
#pragma unroll
          for (int a = 0; a < NA; a++)
            sums[s][a] = __builtin_fma(md, va[a], sums[s][a]);
        }
      }
      __syncthreads();
      staged = next_staged;
    }
  }

  /* final block reduction into a scratch row (like the keyless kernel) */
  const int NV = plan.nslots * (naggs + 1);
  for (int i = tid; i < NV; i += WG) bacc[i] = 0.0;
  __syncthreads();
#pragma unroll
  for (int s = 0; s < NSLOTS; s++) {
    if (s >= plan.nslots) break;
#pragma unroll
    for (int a = 0; a < NA + 1; a++) {
      if (a > naggs) break;
      double x = wave_sum(a < naggs && a < NA ? sums[s][a] : rc[s]);
      if ((tid & 63) == 0 && x != 0.0)
        atomicAdd(&bacc[s * (naggs + 1) + (a < naggs ? a : naggs)], x);
    }
  }
  __syncthreads();
  for (int i = tid; i < NV; i += WG)
    out[(size_t)blockIdx.x * NV + i] = bacc[i];
  (void)out_stride;
}

/* ---- unbounded-cardinality grouped scan (the SHAMap-overflow analogue):
 * accumulates straight into a GLOBAL [nslots][naggs+1] array with f64
 * atomics (zeroed by the host).  High cardinality means low per-address
 * contention, which is exactly when global atomics are cheap; the slot
 * space is still dense (dictionary ids / stats-ranged integers). ---- */
__global__ __launch_bounds__(WG, 1)
void k_grouped_global(sn_dev_plan plan,
                      const sn_dev_plan *__restrict__ plan_g,
                      const sn_dev_batch *__restrict__ batches,
                      const sn_dev_tile *__restrict__ tiles, int ntiles,
                      double *__restrict__ gacc) {
  const int tid = threadIdx.x;
  const int nused = plan.nused;
  const int naggs = plan.naggs, ngroup = plan.ngroup;
  const int npd = plan.npreds_d, npi = plan.npreds_i;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  double *sval = (double *)smem;
  uint64_t *svalid = (uint64_t *)(smem + (size_t)nused * CHUNK * 8);
  uint64_t *sdead = svalid + (size_t)nused * (CHUNK / 64);
  uint64_t *salive = sdead + CHUNK / 64;
  int16_t *sslot = (int16_t *)(salive + CHUNK / 64 + 2);
  sn_dev_plan *P = (sn_dev_plan *)((char *)sslot + CHUNK * 2);
  {
    const GAS unsigned *src = (const GAS unsigned *)(uintptr_t)plan_g;
    unsigned *dst = (unsigned *)P;
    for (unsigned i = tid; i < sizeof(sn_dev_plan) / 4; i += WG) dst[i] = src[i];
  }
  __syncthreads();

  /* 8-way XCD privatization (see the JIT twin): fold happens in k_reduce */
  const int pac = plan.pac;
  const int na1 = pac ? 2 * naggs + 1 : naggs + 1;
  GAS double *acc = (GAS double *)(uintptr_t)gacc +
                    (size_t)(blockIdx.x & 7) * (size_t)plan.nslots * na1;
  const int gc0 = plan.gcol[0], gc1 = plan.gcol[1];
  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + SN_TILE_ROWS, num_rows);
    const int clean = b.clean;

    for (int base = tile.row_start; base < tile_end; base += CHUNK) {
      const int rows = min(CHUNK, tile_end - base);
      convert_chunk(b, nused, base, rows, num_rows, sval, svalid, sdead);
      __syncthreads();
      alive_init(salive, sdead, rows, clean);
      pred_sweeps(P, npd, npi, clean, sval, svalid, salive);
      const int jslot = plan.jkeys && plan.jmode == 1;
      if (plan.jkeys) probe_sweep(P, sval, salive, jslot ? sslot : nullptr);

#pragma unroll 2
      for (int k = 0; k < CHUNK / WG; k++) {
        const int r = tid + k * WG;
        const uint64_t w = salive[r >> 6];
        if (w == 0) continue;
        const int m = (int)((w >> (tid & 63)) & 1ull);
        if (!m) continue;
        long long slot = 0;
        if (jslot) {
          slot = sslot[r];
          if (ngroup >= 1)   /* composite dim_attr x fact_col */
            slot = slot * P->jslot_mul +
                   ((long long)sval[(size_t)gc0 * CHUNK + r] - P->gbase[0]);
        } else {
          if (ngroup >= 1)
            slot = ((long long)sval[(size_t)gc0 * CHUNK + r] -
                    P->gbase[0]) * P->gmul0;
          if (ngroup >= 2)
            slot += (long long)sval[(size_t)gc1 * CHUNK + r] - P->gbase[1];
        }
        GAS double *row_acc = acc + (size_t)slot * na1;
        for (int a = 0; a < naggs; a++) {
          const sn_dev_agg &A = P->aggs[a];
          if (pac) {
            const int av = clean ? 1 : agg_valid(A, svalid, r);
            if (!av) continue;
            (void)atomicAdd((double *)&row_acc[naggs + a], 1.0);
          }
          acc_cell(&row_acc[a], A.op, eval_agg(P, A, sval, r));
        }
        (void)atomicAdd((double *)&row_acc[na1 - 1], 1.0);
      }
      __syncthreads();
    }
  }
}

/* ---- sparse-key open-address hash aggregate ----
 * The ByteBufferHashMap / SHAMapAccessor analogue for integer group keys
 * WITHOUT dense-slot structure (ByteBufferHashMap.putBufferIfAbsent,
 * ByteBufferHashMap.scala:140-183 — open addressing, hash+key probe,
 * append-on-miss; SHAMapAccessor.generateMapGetOrInsert,
 * SHAMapAccessor.scala:716-830).  Instead of the JVM's byte-buffer value
 * records, the device keeps an open-address KEY array in HBM (sentinel
 * SN_HASH_EMPTY, atomicCAS insert) whose probe position IS the accumulator row
 * index — high key cardinality means low per-address contention, exactly
 * when HBM f64 atomics are cheap (same argument as k_grouped_global). */
__device__ __forceinline__ long long sparse_key(const sn_dev_plan *P,
                                                const double *sval, int r) {
  const int gc0 = P->gcol[0];
  if (P->ngroup == 1) {
    const double x = sval[(size_t)gc0 * CHUNK + r];
    return ((P->i64_mask >> gc0) & 1u) ? __double_as_longlong(x)
                                       : (long long)x;
  }
  /* two <=32-bit keys packed into one i64 (reversible; a pack that lands
   * exactly on the sentinel routes to the reserved row like any other) */
  const unsigned k0 = (unsigned)(int)sval[(size_t)gc0 * CHUNK + r];
  const unsigned k1 = (unsigned)(int)sval[(size_t)P->gcol[1] * CHUNK + r];
  return (long long)(((unsigned long long)k0 << 32) | k1);
}

__device__ __forceinline__ int hash_probe(long long key, long long *hk_,
                                          int cap_log2, int32_t *ovf) {
  GAS long long *hk = (GAS long long *)(uintptr_t)hk_;
  if (key == SN_HASH_EMPTY) return 1 << cap_log2;   /* reserved row */
  const unsigned mask = (1u << cap_log2) - 1;
  unsigned h = (unsigned)mix64((unsigned long long)key) & mask;
  for (unsigned it = 0; it <= mask; ++it) {
    const long long k0 = hk[h];
    if (k0 == key) return (int)h;
    if (k0 == SN_HASH_EMPTY) {
      const long long old = (long long)atomicCAS(
          (unsigned long long *)&hk[h], (unsigned long long)SN_HASH_EMPTY,
          (unsigned long long)key);
      if (old == SN_HASH_EMPTY) {
        /* we inserted: count the fill so the host can grow the table
         * BEFORE load factor makes linear probing pathological */
        (void)atomicAdd((int *)(uintptr_t)(ovf + 2), 1);
        return (int)h;
      }
      if (old == key) return (int)h;
      /* lost the insert race to a DIFFERENT key: step on */
    }
    h = (h + 1) & mask;
  }
  atomicOr((int *)(uintptr_t)ovf, 1);               /* table full */
  return -1;
}

__global__ __launch_bounds__(WG, 1)
void k_grouped_hash(sn_dev_plan plan,
                    const sn_dev_plan *__restrict__ plan_g,
                    const sn_dev_batch *__restrict__ batches,
                    const sn_dev_tile *__restrict__ tiles, int ntiles) {
  const int tid = threadIdx.x;
  const int nused = plan.nused;
  const int naggs = plan.naggs;
  const int npd = plan.npreds_d, npi = plan.npreds_i;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  double *sval = (double *)smem;
  uint64_t *svalid = (uint64_t *)(smem + (size_t)nused * CHUNK * 8);
  uint64_t *sdead = svalid + (size_t)nused * (CHUNK / 64);
  uint64_t *salive = sdead + CHUNK / 64;
  sn_dev_plan *P = (sn_dev_plan *)(salive + CHUNK / 64 + 2);
  {
    const GAS unsigned *src = (const GAS unsigned *)(uintptr_t)plan_g;
    unsigned *dst = (unsigned *)P;
    for (unsigned i = tid; i < sizeof(sn_dev_plan) / 4; i += WG) dst[i] = src[i];
  }
  __syncthreads();

  const int pac = plan.pac;
  const int na1 = pac ? 2 * naggs + 1 : naggs + 1;
  GAS double *acc = (GAS double *)(uintptr_t)plan.hacc;
  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + SN_TILE_ROWS, num_rows);
    const int clean = b.clean;

    for (int base = tile.row_start; base < tile_end; base += CHUNK) {
      const int rows = min(CHUNK, tile_end - base);
      convert_chunk(b, nused, base, rows, num_rows, sval, svalid, sdead);
      __syncthreads();
      alive_init(salive, sdead, rows, clean);
      pred_sweeps(P, npd, npi, clean, sval, svalid, salive);
      if (plan.jkeys) probe_sweep(P, sval, salive, nullptr);

#pragma unroll 2
      for (int k = 0; k < CHUNK / WG; k++) {
        const int r = tid + k * WG;
        const uint64_t w = salive[r >> 6];
        if (w == 0) continue;
        const int m = (int)((w >> (tid & 63)) & 1ull);
        if (!m) continue;
        /* a NULL key routes to the dedicated null-group row at cap+1
         * (nullable keys are single-column by the engine contract) */
        int kvalid = 1;
        if (!clean)
          kvalid = (int)((svalid[(size_t)plan.gcol[0] * (CHUNK / 64) +
                                 (r >> 6)] >> (r & 63)) & 1ull);
        int slot;
        if (!kvalid) {
          slot = (1 << plan.hcap_log2) + 1;
        } else {
          const long long key = sparse_key(P, sval, r);
          slot = hash_probe(key, plan.hkeys, plan.hcap_log2, plan.hflags);
          if (slot < 0) continue;                   /* overflow: host retries */
        }
        GAS double *row_acc = acc + (size_t)slot * na1;
        for (int a = 0; a < naggs; a++) {
          const sn_dev_agg &A = P->aggs[a];
          if (pac) {
            const int av = clean ? 1 : agg_valid(A, svalid, r);
            if (!av) continue;
            (void)atomicAdd((double *)&row_acc[naggs + a], 1.0);
          }
          acc_cell(&row_acc[a], A.op, eval_agg(P, A, sval, r));
        }
        (void)atomicAdd((double *)&row_acc[na1 - 1], 1.0);
      }
      __syncthreads();
    }
  }
}

/* ---- device-side join-table build (the partitioned-partitioned /
 * colocated join's HashedObjectCache analogue, HashJoinExec.scala:285-520,
 * ObjectHashSet.scala:107-135): scan the BUILD-side column table's batches
 * and insert (key -> payload gid) into an open-address table with the SAME
 * layout probe_sweep consumes (sentinel LLONG_MIN).  Colocated partitioned
 * tables join bucket-locally in the reference (GemFire colocation), so no
 * exchange step exists on this path. ---- */
__global__ void k_fill_i64(long long *dst, long long n, long long v) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) ((GAS long long *)(uintptr_t)dst)[i] = v;
}

/* flags: [0] = duplicate key with conflicting payload / key == sentinel,
 * [1] = table full */
__global__ __launch_bounds__(WG, 2)
void k_join_build(sn_dev_plan plan,
                  const sn_dev_plan *__restrict__ plan_g,
                  const sn_dev_batch *__restrict__ batches,
                  const sn_dev_tile *__restrict__ tiles, int ntiles,
                  long long *__restrict__ hk_, int32_t *__restrict__ hp_,
                  int cap_log2, int has_attr, int32_t *__restrict__ flags_) {
  const int tid = threadIdx.x;
  const int nused = plan.nused;
  GAS long long *hk = (GAS long long *)(uintptr_t)hk_;
  GAS int32_t *hp = (GAS int32_t *)(uintptr_t)hp_;
  GAS int32_t *flags = (GAS int32_t *)(uintptr_t)flags_;
  const unsigned mask = (1u << cap_log2) - 1;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  double *sval = (double *)smem;
  uint64_t *svalid = (uint64_t *)(smem + (size_t)nused * CHUNK * 8);
  uint64_t *sdead = svalid + (size_t)nused * (CHUNK / 64);
  uint64_t *salive = sdead + CHUNK / 64;

  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + SN_TILE_ROWS, num_rows);
    const int clean = b.clean;

    for (int base = tile.row_start; base < tile_end; base += CHUNK) {
      const int rows = min(CHUNK, tile_end - base);
      convert_chunk(b, nused, base, rows, num_rows, sval, svalid, sdead);
      __syncthreads();
      alive_init(salive, sdead, rows, clean);
      if (!clean) {
        /* null build keys never join (SQL equality) */
        const int tid64 = tid & 63;
        (void)tid64;
#pragma unroll
        for (int k = 0; k < CHUNK / WG; k++) {
          const int r = tid + k * WG;
          uint64_t w = svalid[(size_t)0 * (CHUNK / 64) + (r >> 6)];
          if ((tid & 63) == 0) salive[r >> 6] &= w;
        }
      }
#pragma unroll 2
      for (int k = 0; k < CHUNK / WG; k++) {
        const int r = tid + k * WG;
        const uint64_t w = salive[r >> 6];
        if (w == 0) continue;
        if (!((w >> (tid & 63)) & 1ull)) continue;
        const double kx = sval[r];
        const long long key = ((plan.i64_mask >> 0) & 1u)
                                  ? __double_as_longlong(kx)
                                  : (long long)kx;
        const int32_t pay = has_attr ? (int32_t)sval[(size_t)1 * CHUNK + r] : 0;
        if (key == LLONG_MIN) { atomicOr((int *)&flags[0], 1); continue; }
        unsigned h = (unsigned)mix64((unsigned long long)key) & mask;
        int done = 0;
        /* payload publication: the inserter CASes the key then atomically
         * exchanges the payload over the INT32_MIN sentinel; a concurrent
         * duplicate spins (bounded) until the payload is visible before
         * checking for a conflict */
        for (unsigned it = 0; it <= mask && !done; ++it) {
          long long k0 = hk[h];
          if (k0 == LLONG_MIN) {
            k0 = (long long)atomicCAS(
                (unsigned long long *)&hk[h], (unsigned long long)LLONG_MIN,
                (unsigned long long)key);
            if (k0 == LLONG_MIN) {              /* we inserted */
              (void)atomicExch((int *)&hp[h], pay);
              done = 1;
              break;
            }
          }
          if (k0 == key) {
            int pv;
            int spins = 0;
            do {
              pv = atomicOr((int *)&hp[h], 0);  /* atomic read */
            } while (pv == INT_MIN && ++spins < (1 << 20));
            if (pv != pay) atomicOr((int *)&flags[0], 1);
            done = 1;
            break;
          }
          h = (h + 1) & mask;
        }
        if (!done) atomicOr((int *)&flags[1], 1);
      }
      __syncthreads();
    }
  }
}

/* densify the open-address table into a payload LUT over [lmin, lmax] */
__global__ void k_hash_to_lut(const long long *__restrict__ hk_,
                              const int32_t *__restrict__ hp_,
                              long long cap, int32_t *__restrict__ lut_,
                              long long lmin) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= cap) return;
  const long long key = ((const GAS long long *)(uintptr_t)hk_)[i];
  if (key == LLONG_MIN) return;
  ((GAS int32_t *)(uintptr_t)lut_)[key - lmin] =
      ((const GAS int32_t *)(uintptr_t)hp_)[i];
}

extern "C" int sn_launch_join_build(const sn_dev_plan *plan,
                                    const sn_dev_plan *dev_plan,
                                    const sn_dev_batch *dev_batches,
                                    const sn_dev_tile *dev_tiles,
                                    int32_t ntiles, long long *hk,
                                    int32_t *hp, int cap_log2, int has_attr,
                                    int32_t *flags, void *stream) {
  hipStream_t s = (hipStream_t)stream;
  {
    long long cap = 1ll << cap_log2;
    hipLaunchKernelGGL(k_fill_i64, dim3((unsigned)((cap + 255) / 256)),
                       dim3(256), 0, s, hk, cap, LLONG_MIN);
    /* payloads init to the INT32_MIN publication sentinel (two per i64) */
    hipLaunchKernelGGL(k_fill_i64, dim3((unsigned)((cap / 2 + 255) / 256)),
                       dim3(256), 0, s, (long long *)hp, cap / 2,
                       (long long)0x8000000080000000ull);
  }
  int grid;
  if (ntiles <= 0) return (int)hipGetLastError();
  if (ntiles <= SN_GRID_CAP) grid = ntiles;
  else {
    int rounds = (ntiles + SN_GRID_CAP - 1) / SN_GRID_CAP;
    grid = (ntiles + rounds - 1) / rounds;
  }
  size_t lds = (size_t)plan->nused * CHUNK * 8 +
               (size_t)plan->nused * (CHUNK / 64) * 8 +
               2 * (CHUNK / 64) * 8 + 64;
  if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
  hipLaunchKernelGGL(k_join_build, dim3(grid), dim3(WG), lds, s,
                     *plan, dev_plan, dev_batches, dev_tiles, ntiles,
                     hk, hp, cap_log2, has_attr, flags);
  return (int)hipGetLastError();
}

extern "C" int sn_launch_hash_to_lut(const long long *hk, const int32_t *hp,
                                     long long cap, int32_t *lut,
                                     long long lmin, void *stream) {
  hipLaunchKernelGGL(k_hash_to_lut, dim3((unsigned)((cap + 255) / 256)),
                     dim3(256), 0, (hipStream_t)stream, hk, hp, cap, lut, lmin);
  return (int)hipGetLastError();
}

/* compact the used hash-table rows into dense (key, accumulator-row) pairs
 * so the host reads back only the live groups, not the whole table */
__global__ void k_hash_compact(const long long *__restrict__ hk_,
                               const double *__restrict__ hacc_,
                               int cap, int naggs1,
                               long long *__restrict__ okeys,
                               double *__restrict__ orows,
                               int *__restrict__ counter) {
  const GAS long long *hk = (const GAS long long *)(uintptr_t)hk_;
  const GAS double *hacc = (const GAS double *)(uintptr_t)hacc_;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i > cap) return;
  long long key;
  int used;
  if (i == cap) {                                   /* reserved sentinel-key row */
    key = SN_HASH_EMPTY;
    used = hacc[(size_t)cap * naggs1 + (naggs1 - 1)] != 0.0;
  } else {
    key = hk[i];
    used = key != SN_HASH_EMPTY;
  }
  if (!used) return;
  const int o = atomicAdd((int *)(uintptr_t)counter, 1);
  ((GAS long long *)(uintptr_t)okeys)[o] = key;
  GAS double *dst = (GAS double *)(uintptr_t)orows + (size_t)o * naggs1;
  for (int a = 0; a < naggs1; a++) dst[a] = hacc[(size_t)i * naggs1 + a];
}

extern "C" int sn_launch_hash_scan(const sn_dev_plan *plan,
                                   const sn_dev_plan *dev_plan,
                                   const sn_dev_batch *dev_batches,
                                   const sn_dev_tile *dev_tiles, int32_t ntiles,
                                   void *stream) {
  hipStream_t s = (hipStream_t)stream;
  int grid;
  if (ntiles <= 0) return 0;
  if (ntiles <= SN_GRID_CAP) grid = ntiles;
  else {
    int rounds = (ntiles + SN_GRID_CAP - 1) / SN_GRID_CAP;
    grid = (ntiles + rounds - 1) / rounds;
  }
  size_t lds = (size_t)plan->nused * CHUNK * 8 +
               (size_t)plan->nused * (CHUNK / 64) * 8 +
               2 * (CHUNK / 64) * 8 + sizeof(sn_dev_plan) + 64;
  if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
  hipLaunchKernelGGL(k_grouped_hash, dim3(grid), dim3(WG), lds, s,
                     *plan, dev_plan, dev_batches, dev_tiles, ntiles);
  return (int)hipGetLastError();
}

extern "C" int sn_launch_hash_compact(const long long *hk, const double *hacc,
                                      int cap, int naggs1, long long *okeys,
                                      double *orows, int *counter,
                                      void *stream) {
  hipLaunchKernelGGL(k_hash_compact, dim3((cap + 1 + 255) / 256), dim3(256), 0,
                     (hipStream_t)stream, hk, hacc, cap, naggs1, okeys, orows,
                     counter);
  return (int)hipGetLastError();
}

/* ---- radix pass 2: per-partition aggregation in LDS ----
 * Pass 1 (the query-compiled scatter) partitioned the passing rows by
 * mix64(key) high bits into plan->precs.  Here one workgroup owns one
 * partition and aggregates its records into an LDS-RESIDENT open-address
 * table (the measured fix: a first version kept per-partition segments of
 * the global table — "L2-resident" probes still cost 6.9 ms on 60M rows
 * at 1 workgroup/partition, nearly the whole single-pass time, because
 * the streaming record reads evict the segments and the dependent L2
 * atomic chains sit at 8 waves/CU; LDS f64 atomics are the same pipe the
 * 5 TB/s dense grouped kernels run on).  The filled slots then compact
 * DIRECTLY from LDS into the (okeys, orows) result arrays — the global
 * hkeys table is never touched and k_hash_compact is skipped; only the
 * reserved sentinel-key row still lives in hacc[cap].  Fill counting and
 * the overflow flag keep the single-pass contract, so grow-and-retry is
 * unchanged. */
__global__ __launch_bounds__(WG, 1)
void k_radix_agg_lds(const sn_dev_plan *__restrict__ plan_g,
                     long long *__restrict__ okeys_,
                     double *__restrict__ orows_,
                     int *__restrict__ counter_) {
  extern __shared__ __attribute__((aligned(16))) char lsm[];
  const GAS sn_dev_plan *P = (const GAS sn_dev_plan *)(uintptr_t)plan_g;
  const int naggs = P->naggs;
  const int na1 = naggs + 1;              /* radix requires pac == 0 */
  const int sub = P->radix;               /* partition shift (slots log2) */
  const int subcap = 1 << sub;
  const unsigned smask = (unsigned)subcap - 1;
  const int tid = threadIdx.x;
  const int p = blockIdx.x;
  long long *skeys = (long long *)lsm;
  double *sacc = (double *)(lsm + (size_t)subcap * 8);
  int *scnt = (int *)(lsm + (size_t)subcap * 8 + (size_t)subcap * na1 * 8);
  __shared__ int basew;

  int ops[12];                            /* SN_MAX_AGGS */
  double idents[12];
  for (int a = 0; a < naggs; a++) {
    ops[a] = P->aggs[a].op;
    idents[a] = acc_ident(ops[a]);
  }
  for (int i = tid; i < subcap; i += WG) {
    skeys[i] = SN_HASH_EMPTY;
    double *row = sacc + (size_t)i * na1;
    for (int a = 0; a < naggs; a++) row[a] = idents[a];
    row[na1 - 1] = 0.0;
  }
  __syncthreads();

  const int percap = P->percap;
  const int n = min(((const GAS int *)(uintptr_t)P->pcount)[p], percap);
  const GAS double *recs = (const GAS double *)(uintptr_t)P->precs +
                           (size_t)p * percap * (1 + naggs);
  GAS double *acc = (GAS double *)(uintptr_t)P->hacc;
  GAS int32_t *flags = (GAS int32_t *)(uintptr_t)P->hflags;
  const int cap = 1 << P->hcap_log2;

  for (int i = tid; i < n; i += WG) {
    const GAS double *rec = recs + (size_t)i * (1 + naggs);
    const long long key = __double_as_longlong(rec[0]);
    if (key == SN_HASH_EMPTY) {
      /* a REAL key of -1: reserved global row at cap (rare) */
      GAS double *row = acc + (size_t)cap * na1;
      for (int a = 0; a < naggs; a++) acc_cell(&row[a], ops[a], rec[1 + a]);
      (void)atomicAdd((double *)&row[na1 - 1], 1.0);
      continue;
    }
    unsigned h = (unsigned)mix64((unsigned long long)key) & smask;
    int slot = -1;
    for (unsigned it = 0; it <= smask; ++it) {
      const long long k0 = skeys[h];
      if (k0 == key) { slot = (int)h; break; }
      if (k0 == SN_HASH_EMPTY) {
        const long long old = (long long)atomicCAS(
            (unsigned long long *)&skeys[h], (unsigned long long)SN_HASH_EMPTY,
            (unsigned long long)key);
        if (old == SN_HASH_EMPTY || old == key) { slot = (int)h; break; }
      }
      h = (h + 1) & smask;
    }
    if (slot < 0) { atomicOr((int *)flags, 1); continue; }   /* table full */
    double *row = sacc + (size_t)slot * na1;
    for (int a = 0; a < naggs; a++) acc_cell(&row[a], ops[a], rec[1 + a]);
    (void)atomicAdd(&row[na1 - 1], 1.0);
  }
  __syncthreads();

  /* compact straight from LDS: thread t owns slots [t*spt, (t+1)*spt);
   * block prefix-sum places each filled slot at one reserved output
   * range (ONE global counter add per block, and the filled-slot total
   * IS the insert count, so the fill counter costs one more add). */
  const int spt = subcap / WG;
  int mine = 0;
  for (int j = 0; j < spt; j++)
    mine += skeys[tid * spt + j] != SN_HASH_EMPTY;
  scnt[tid] = mine;
  __syncthreads();
  for (int off = 1; off < WG; off <<= 1) {        /* inclusive scan */
    const int u = tid >= off ? scnt[tid - off] : 0;
    __syncthreads();
    scnt[tid] += u;
    __syncthreads();
  }
  if (tid == WG - 1) {
    const int total = scnt[WG - 1];
    basew = total ? atomicAdd((int *)(uintptr_t)counter_, total) : 0;
    if (total) (void)atomicAdd((int *)(flags + 2), total);
  }
  __syncthreads();
  int o = basew + scnt[tid] - mine;
  GAS long long *okeys = (GAS long long *)(uintptr_t)okeys_;
  GAS double *orows = (GAS double *)(uintptr_t)orows_;
  for (int j = 0; j < spt; j++) {
    const int s = tid * spt + j;
    if (skeys[s] == SN_HASH_EMPTY) continue;
    okeys[o] = skeys[s];
    GAS double *dst = orows + (size_t)o * na1;
    const double *src = sacc + (size_t)s * na1;
    for (int a = 0; a < na1; a++) dst[a] = src[a];
    o++;
  }
}

/* fallback for accumulator rows too wide for the LDS table: per-partition
 * SEGMENTS of the global hkeys/hacc (compact + readback then run as in the
 * single-pass path) */
__global__ __launch_bounds__(WG, 2)
void k_radix_agg_glob(const sn_dev_plan *__restrict__ plan_g) {
  const GAS sn_dev_plan *P = (const GAS sn_dev_plan *)(uintptr_t)plan_g;
  const int naggs = P->naggs;
  const int na1 = naggs + 1;
  const int sub = P->radix;
  const int p = blockIdx.x;
  const int percap = P->percap;
  const GAS int *pcount = (const GAS int *)(uintptr_t)P->pcount;
  const int n = min(pcount[p], percap);
  const GAS double *recs = (const GAS double *)(uintptr_t)P->precs +
                           (size_t)p * percap * (1 + naggs);
  GAS long long *hk = (GAS long long *)(uintptr_t)P->hkeys;
  GAS double *acc = (GAS double *)(uintptr_t)P->hacc;
  GAS int32_t *flags = (GAS int32_t *)(uintptr_t)P->hflags;
  const int cap = 1 << P->hcap_log2;
  const int segbase = p << sub;
  const unsigned smask = (1u << sub) - 1;
  int ops[12];
  for (int a = 0; a < naggs; a++) ops[a] = P->aggs[a].op;

  for (int i = threadIdx.x; i < n; i += WG) {
    const GAS double *rec = recs + (size_t)i * (1 + naggs);
    const long long key = __double_as_longlong(rec[0]);
    int slot;
    if (key == SN_HASH_EMPTY) {
      slot = cap;                                   /* reserved row */
    } else {
      unsigned h = (unsigned)mix64((unsigned long long)key) & smask;
      slot = -1;
      for (unsigned it = 0; it <= smask; ++it) {
        const int s = segbase + (int)h;
        const long long k0 = hk[s];
        if (k0 == key) { slot = s; break; }
        if (k0 == SN_HASH_EMPTY) {
          const long long old = (long long)atomicCAS(
              (unsigned long long *)&hk[s], (unsigned long long)SN_HASH_EMPTY,
              (unsigned long long)key);
          if (old == SN_HASH_EMPTY) {
            (void)atomicAdd((int *)(flags + 2), 1);
            slot = s; break;
          }
          if (old == key) { slot = s; break; }
        }
        h = (h + 1) & smask;
      }
      if (slot < 0) { atomicOr((int *)flags, 1); continue; }  /* segment full */
    }
    GAS double *row = acc + (size_t)slot * na1;
    for (int a = 0; a < naggs; a++) acc_cell(&row[a], ops[a], rec[1 + a]);
    (void)atomicAdd((double *)&row[na1 - 1], 1.0);
  }
}

extern "C" int sn_launch_radix_agg(const sn_dev_plan *plan,
                                   const sn_dev_plan *dev_plan,
                                   long long *okeys, double *orows,
                                   int *counter, void *stream) {
  const int npart = 1 << (plan->hcap_log2 - plan->radix);
  const int na1 = plan->naggs + 1;
  if (sn_radix_direct(plan->radix, na1)) {
    const size_t lds = sn_radix_lds_bytes(plan->radix, na1);
    hipLaunchKernelGGL(k_radix_agg_lds, dim3(npart), dim3(WG), lds,
                       (hipStream_t)stream, dev_plan, okeys, orows, counter);
  } else {
    hipLaunchKernelGGL(k_radix_agg_glob, dim3(npart), dim3(WG), 0,
                       (hipStream_t)stream, dev_plan);
  }
  return (int)hipGetLastError();
}

/* ---- device-side batch stats (f2: GPU batch building) ----
 * min/max over one raw fixed-width column (the ColumnStatsSchema bounds
 * ColumnEncoder tracks during encode, ColumnEncoding.scala:188-251),
 * accumulated into out[0]=min, out[1]=max with ord-encoded u64 atomics
 * (doubles) or native i64 atomics — async on the put stream, so ingest
 * never syncs per batch. */
__device__ __forceinline__ unsigned long long wave_umin(unsigned long long x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const unsigned long long y =
        (unsigned long long)__shfl_down((long long)x, off, 64);
    x = y < x ? y : x;
  }
  return x;
}
__device__ __forceinline__ unsigned long long wave_umax(unsigned long long x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const unsigned long long y =
        (unsigned long long)__shfl_down((long long)x, off, 64);
    x = y > x ? y : x;
  }
  return x;
}

/* omin slots memset to 0xFF (>= every encoding), omax to 0x00: f64 values
 * use the ord encoding, integers the sign-bias (v ^ 1<<63) — both
 * order-preserving into unsigned, so the memset identities work for all */
__global__ __launch_bounds__(256, 4)
void k_col_minmax(const void *__restrict__ body_, long long n, int kind,
                  unsigned long long *__restrict__ omin_,
                  unsigned long long *__restrict__ omax_) {
  GAS unsigned long long *omin = (GAS unsigned long long *)(uintptr_t)omin_;
  GAS unsigned long long *omax = (GAS unsigned long long *)(uintptr_t)omax_;
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  unsigned long long emn = ~0ull, emx = 0ull;
  if (kind == SN_K_F64 || kind == SN_K_F32) {
    double mn = __longlong_as_double(0x7FF0000000000000ll);   /* +inf */
    double mx = -mn;
    bool any = false;
    if (kind == SN_K_F64) {
      const GAS double *b = (const GAS double *)(uintptr_t)body_;
      for (long long i = i0; i < n; i += stride) {
        const double v = b[i];
        mn = fmin(mn, v); mx = fmax(mx, v); any = true;
      }
    } else {
      const GAS float *b = (const GAS float *)(uintptr_t)body_;
      for (long long i = i0; i < n; i += stride) {
        const double v = (double)b[i];
        mn = fmin(mn, v); mx = fmax(mx, v); any = true;
      }
    }
    if (any) { emn = f64_ord(mn); emx = f64_ord(mx); }
  } else {
    long long mn = 0x7FFFFFFFFFFFFFFFll, mx = (long long)0x8000000000000000ll;
    bool any = false;
    if (kind == SN_K_I64) {
      const GAS long long *b = (const GAS long long *)(uintptr_t)body_;
      for (long long i = i0; i < n; i += stride) {
        const long long v = b[i];
        mn = v < mn ? v : mn; mx = v > mx ? v : mx; any = true;
      }
    } else if (kind == SN_K_I32) {
      const GAS int *b = (const GAS int *)(uintptr_t)body_;
      for (long long i = i0; i < n; i += stride) {
        const long long v = b[i];
        mn = v < mn ? v : mn; mx = v > mx ? v : mx; any = true;
      }
    } else {  /* I16 */
      const GAS short *b = (const GAS short *)(uintptr_t)body_;
      for (long long i = i0; i < n; i += stride) {
        const long long v = b[i];
        mn = v < mn ? v : mn; mx = v > mx ? v : mx; any = true;
      }
    }
    if (any) {
      emn = (unsigned long long)mn ^ 0x8000000000000000ull;
      emx = (unsigned long long)mx ^ 0x8000000000000000ull;
    }
  }
  emn = wave_umin(emn);
  emx = wave_umax(emx);
  if ((threadIdx.x & 63) == 0) {
    (void)atomicMin((unsigned long long *)omin, emn);
    (void)atomicMax((unsigned long long *)omax, emx);
  }
}

extern "C" int sn_launch_col_minmax(const void *body, long long n, int kind,
                                    unsigned long long *omin,
                                    unsigned long long *omax, void *stream) {
  long long blocks = (n + 256 * 16 - 1) / (256 * 16);
  if (blocks < 1) blocks = 1;
  if (blocks > 1024) blocks = 1024;
  hipLaunchKernelGGL(k_col_minmax, dim3((unsigned)blocks), dim3(256), 0,
                     (hipStream_t)stream, body, n, kind, omin, omax);
  return (int)hipGetLastError();
}

/* ---- wave-cooperative LZ4 block decode (f1: compressed-upload ingest) ----
 * The reference wraps each column blob as ONE raw LZ4 block
 * (CompressionUtils.scala:132-160): the token chain is strictly
 * sequential, so lane 0 parses while all 64 lanes of the wave copy the
 * literal/match runs — wave lockstep makes the copy loop barrier-free.
 * Overlapping matches (offset < length) replicate periodically:
 * dst[i] = dst[start - off + i % off], every read landing before the
 * segment start.  One wave per blob; concurrency comes from decoding
 * many blobs in flight (one launch per put on the engine stream). */
__global__ __launch_bounds__(64, 8)
void k_lz4_decompress(const uint8_t *__restrict__ src_, long long slen,
                      uint8_t *__restrict__ dst_, long long dlen,
                      int32_t *__restrict__ err_) {
  const GAS uint8_t *src = (const GAS uint8_t *)(uintptr_t)src_;
  GAS uint8_t *dst = (GAS uint8_t *)(uintptr_t)dst_;
  GAS int32_t *err = (GAS int32_t *)(uintptr_t)err_;
  const int lane = threadIdx.x;
  long long ip = 0, op = 0;
  while (true) {
    /* lane 0 parses one sequence; broadcast (lit_src, lit_len, m_off,
     * m_len, next_ip); next_ip -2 = clean end, -3 = malformed */
    long long v0 = 0, v1 = 0, v2 = 0, v3 = 0, v4 = -3;
    if (lane == 0) {
      if (ip >= slen) {
        v4 = -2;
      } else {
        const unsigned tok = src[ip++];
        long long ll = tok >> 4;
        if (ll == 15) {
          unsigned b;
          do { b = (ip < slen) ? src[ip++] : 0; ll += b; }
          while (b == 255 && ip < slen);
        }
        v0 = ip;                                /* literal source */
        v1 = ll;
        ip += ll;
        if (ip > slen || op + ll > dlen) {
          v4 = -3;
        } else if (ip == slen) {
          v4 = ip;                              /* final literals-only seq */
        } else if (ip + 2 > slen) {
          v4 = -3;
        } else {
          long long off = (long long)src[ip] | ((long long)src[ip + 1] << 8);
          ip += 2;
          long long ml = tok & 15;
          if (ml == 15) {
            unsigned b;
            do { b = (ip < slen) ? src[ip++] : 0; ml += b; }
            while (b == 255 && ip < slen);
          }
          ml += 4;
          if (off == 0 || off > op + ll || op + ll + ml > dlen) v4 = -3;
          else { v2 = off; v3 = ml; v4 = ip; }
        }
      }
    }
    v0 = __shfl(v0, 0, 64);
    v1 = __shfl(v1, 0, 64);
    v2 = __shfl(v2, 0, 64);
    v3 = __shfl(v3, 0, 64);
    v4 = __shfl(v4, 0, 64);
    if (v4 == -2) break;                        /* end of input */
    if (v4 == -3) { if (lane == 0 && err) err[0] = 1; return; }
    for (long long i = lane; i < v1; i += 64) dst[op + i] = src[v0 + i];
    op += v1;
    if (v3 > 0) {
      /* match copy: direct when the source window clears the segment,
       * periodic replication when it overlaps (off < len) — every read
       * lands strictly before the segment start either way */
      const long long mo = v2;
      for (long long i = lane; i < v3; i += 64)
        dst[op + i] = dst[op - mo + (mo >= v3 ? i : i % mo)];
      op += v3;
    }
    ip = v4;
  }
  if (lane == 0 && err) err[0] = (op == dlen) ? 0 : 2;
}

extern "C" int sn_launch_lz4_decompress(const void *src, long long slen,
                                        void *dst, long long dlen,
                                        int32_t *err, void *stream) {
  hipLaunchKernelGGL(k_lz4_decompress, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const uint8_t *)src, slen,
                     (uint8_t *)dst, dlen, err);
  return (int)hipGetLastError();
}

/* fold per-block partial rows into the final output.
 * keyless: final[i] = sum_b scratch[b][i]  (NV = 2*NA_t+1, identical layout)
 * grouped: scratch rows are [slot][na1]; final is [slot][out_stride]
 * with rowcount at out_stride-1.  MIN/MAX cells (op from plan_g, NULL =
 * all-sum) hold the ord-u64 encoding in the scratch rows; they fold with
 * integer min/max and DECODE to plain doubles here. */
__device__ __forceinline__ double ord_f64(unsigned long long o) {
  unsigned long long b = (o & 0x8000000000000000ull)
      ? (o ^ 0x8000000000000000ull) : ~o;
  return __longlong_as_double((long long)b);
}

__global__ void k_reduce(const double *__restrict__ scratch, int nblocks,
                         int nv, double *__restrict__ out, int naggs1,
                         int out_stride,
                         const sn_dev_plan *__restrict__ plan_g) {
  /* one block per output value; 256 threads stride the partial rows */
  __shared__ double red[4];
  __shared__ unsigned long long redu[4];
  const GAS double *src = (const GAS double *)(uintptr_t)scratch;
  const int i = blockIdx.x;
  if (i >= nv) return;
  int op = 0;
  if (plan_g && naggs1 > 0) {
    const GAS sn_dev_plan *P = (const GAS sn_dev_plan *)(uintptr_t)plan_g;
    const int naggs = P->naggs;
    const int a = i % naggs1;
    if (a < naggs) op = P->aggs[a].op;
  }
  if (op != 0) {
    unsigned long long m = op == 1 ? ORD_MIN_IDENT : ORD_MAX_IDENT;
    for (int b = threadIdx.x; b < nblocks; b += blockDim.x) {
      const unsigned long long v =
          (unsigned long long)__double_as_longlong(src[(size_t)b * nv + i]);
      m = op == 1 ? (v < m ? v : m) : (v > m ? v : m);
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const unsigned long long o2 =
          (unsigned long long)__shfl_down((long long)m, off, 64);
      m = op == 1 ? (o2 < m ? o2 : m) : (o2 > m ? o2 : m);
    }
    if ((threadIdx.x & 63) == 0) redu[threadIdx.x >> 6] = m;
    __syncthreads();
    if (threadIdx.x == 0) {
      for (int w = 1; w < 4; w++)
        m = op == 1 ? (redu[w] < redu[0] ? redu[w] : redu[0])
                    : (redu[w] > redu[0] ? redu[w] : redu[0]),
        redu[0] = m;
      const int slot = i / naggs1, a = i % naggs1;
      out[(size_t)slot * out_stride + (a < naggs1 - 1 ? a : out_stride - 1)] =
          ord_f64(redu[0]);
    }
    return;
  }
  double s = 0.0;
  for (int b = threadIdx.x; b < nblocks; b += blockDim.x)
    s += src[(size_t)b * nv + i];
  s = wave_sum(s);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    double t = red[0] + red[1] + red[2] + red[3];
    if (naggs1 == 0) {
      out[i] = t;
    } else {
      int slot = i / naggs1, a = i % naggs1;
      out[(size_t)slot * out_stride + (a < naggs1 - 1 ? a : out_stride - 1)] = t;
    }
  }
}

/* initialize a global accumulator's MIN/MAX cells to their identities
 * (host memset covers only the sum/count cells' zero) */
__global__ void k_acc_init(double *__restrict__ acc, long long rows,
                           int naggs, int na1,
                           const sn_dev_plan *__restrict__ plan_g) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= rows * na1) return;
  const int a = (int)(i % na1);
  if (a >= naggs) return;
  const GAS sn_dev_plan *P = (const GAS sn_dev_plan *)(uintptr_t)plan_g;
  const int op = P->aggs[a].op;
  if (op == 0) return;
  ((GAS double *)(uintptr_t)acc)[i] = acc_ident(op);
}

extern "C" int sn_launch_acc_init(double *acc, long long rows, int naggs,
                                  int na1, const void *plan_dev,
                                  void *stream) {
  long long n = rows * na1;
  hipLaunchKernelGGL(k_acc_init, dim3((unsigned)((n + 255) / 256)), dim3(256),
                     0, (hipStream_t)stream, acc, rows, naggs, na1,
                     (const sn_dev_plan *)plan_dev);
  return (int)hipGetLastError();
}

/* put-time patch materialization: write value-only update patches straight
 * into the (null-free, fixed-width) device body so the batch scans clean.
 * Values are plain doubles (decode_delta widening) except INT64, whose
 * patch values travel as raw bits (exact beyond 2^53); the narrower integer
 * bodies take round-nearest, matching read_general's __double2ll_rn. */
__global__ void k_patch_apply(void *__restrict__ body,
                              const int32_t *__restrict__ pos,
                              const double *__restrict__ val, int n,
                              int kind) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const int p = as_global(pos)[i];
  const double v = as_global(val)[i];
  switch (kind) {
    case SN_K_F64: ((GAS double *)(uintptr_t)body)[p] = v; break;
    case SN_K_F32: ((GAS float *)(uintptr_t)body)[p] = (float)v; break;
    case SN_K_I32: ((GAS int32_t *)(uintptr_t)body)[p] = (int32_t)__double2ll_rn(v); break;
    /* INT64 patch values are raw bits (exact beyond 2^53) */
    case SN_K_I64: ((GAS long long *)(uintptr_t)body)[p] = __double_as_longlong(v); break;
    case SN_K_I16: ((GAS int16_t *)(uintptr_t)body)[p] = (int16_t)__double2ll_rn(v); break;
  }
}

extern "C" int sn_launch_patch_apply(void *body, const int32_t *pos,
                                     const double *val, int n, int kind,
                                     void *stream) {
  if (n <= 0) return 0;
  hipLaunchKernelGGL(k_patch_apply, dim3((n + 255) / 256), dim3(256), 0,
                     (hipStream_t)stream, body, pos, val, n, kind);
  return (int)hipGetLastError();
}

extern "C" int sn_launch_reduce(const double *dev_scratch, int nblocks,
                                int nv, double *dev_out, int naggs1,
                                int out_stride, const void *plan_dev,
                                void *stream) {
  hipLaunchKernelGGL(k_reduce, dim3(nv), dim3(WG), 0, (hipStream_t)stream,
                     dev_scratch, nblocks, nv, dev_out, naggs1, out_stride,
                     (const sn_dev_plan *)plan_dev);
  return (int)hipGetLastError();
}

extern "C" int sn_launch_scan_agg(const sn_dev_plan *plan,
                                  const sn_dev_plan *dev_plan,
                                  const sn_dev_batch *dev_batches,
                                  const sn_dev_tile *dev_tiles, int32_t ntiles,
                                  double *dev_out, double *dev_scratch,
                                  void *stream) {
  hipStream_t s = (hipStream_t)stream;
  /* balance: every block gets the same tile count (a ragged grid-stride
   * leaves half the blocks with 2x work and CUs idle in the tail) */
  int grid;
  if (ntiles <= 0) grid = 1;
  else if (ntiles <= SN_GRID_CAP) grid = ntiles;
  else {
    int rounds = (ntiles + SN_GRID_CAP - 1) / SN_GRID_CAP;
    grid = (ntiles + rounds - 1) / rounds;
  }
  const int ns = plan->nslots, na = plan->naggs;
  const int na_t = na <= 2 ? 2 : na <= 4 ? 4 : na <= 8 ? 8 : 12;
  size_t lds = (size_t)plan->nused * CHUNK * 8 +
               (size_t)plan->nused * (CHUNK / 64) * 8 +
               2 * (CHUNK / 64) * 8 +            /* sdead + salive */
               sizeof(sn_dev_plan) + 512;        /* plan mirror + keyless bacc */
  hipError_t err;
  const bool nc4 = plan->nused <= 4;
  const int na1 = plan->pac ? 2 * na + 1 : na + 1;
  int nv, naggs1, out_stride;
  if (ns <= 1 && !plan->pac) {
    nv = 2 * na_t + 1; naggs1 = 0; out_stride = nv;
  } else {
    /* grouped layout (also keyless MIN/MAX plans routed through the
     * grouped kernels with one slot) */
    nv = (ns < 1 ? 1 : ns) * na1; naggs1 = na1; out_stride = 2 * na_t + 1;
  }
#define KL(A, NCv) hipLaunchKernelGGL((k_keyless<A, NCv>), dim3(grid), dim3(WG), lds, s, \
        *plan, dev_plan, dev_batches, dev_tiles, ntiles, dev_scratch)
#define KG(S, NCv) hipLaunchKernelGGL((k_grouped<S, NCv>), dim3(grid), dim3(WG), lds, s, \
        *plan, dev_plan, dev_batches, dev_tiles, ntiles, dev_scratch, out_stride)
  if (ns <= 1 && !plan->pac) {
    if (na <= 2) { if (nc4) KL(2, 4); else KL(2, 8); }
    else if (na <= 4) { if (nc4) KL(4, 4); else KL(4, 8); }
    else { if (nc4) KL(12, 4); else KL(12, 8); }
  } else if (ns > SN_RESULT_PAGE ||
             (plan->pac &&
              sn_grouped_needs_global(plan->nused, ns, na1))) {
    /* unbounded cardinality (or a pac accumulator too wide for LDS):
     * global f64 atomics into the (host-zeroed) scratch accumulator */
    lds += (CHUNK / 64) * 8 + CHUNK * 2 + sizeof(sn_dev_plan) + 64;
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    hipLaunchKernelGGL(k_grouped_global, dim3(grid), dim3(WG), lds, s,
                       *plan, dev_plan, dev_batches, dev_tiles, ntiles,
                       dev_scratch);
    grid = 8;   /* k_reduce folds the 8 XCD-private copies */
  } else if (ns > 16 || plan->pac) {
    /* large-cardinality LDS hash-aggregate path (also the per-agg-count
     * path for nullable aggregate inputs at any slot count): the LDS
     * accumulator bounds the grid so scratch rows stay small */
    if (grid > SN_GRID_BIGSLOT) grid = SN_GRID_BIGSLOT;
    lds += (CHUNK / 64) * 8 + CHUNK * 2 + (size_t)ns * na1 * 8 + 64;
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    hipLaunchKernelGGL(k_grouped_lds, dim3(grid), dim3(WG), lds, s,
                       *plan, dev_plan, dev_batches, dev_tiles, ntiles,
                       dev_scratch, out_stride);
  } else if (plan->jmode != 1 && ns <= 8 && na <= 6) {
    /* register-accumulator grouped kernel (Q1's shape) */
    lds += (CHUNK / 64) * 8 + (size_t)8 * (na + 1) * 8 + 64;
#define KGR(S, A, NCv) hipLaunchKernelGGL((k_grouped_reg<S, A, NCv>), dim3(grid), \
        dim3(WG), lds, s, *plan, dev_plan, dev_batches, dev_tiles, ntiles, \
        dev_scratch, out_stride)
    /* (measured: an unstaged 3-waves/SIMD variant loses 20% — staging
     * beats occupancy for the grouped shape as well) */
    if (nc4) KGR(8, 6, 4); else KGR(8, 6, 8);
#undef KGR
  } else {
    /* bacc + plan-mirror offsets inside the kernel use the TEMPLATE slot
     * count — size the dynamic LDS with it, not the runtime ns */
    const int ns_t = ns <= 4 ? 4 : ns <= 8 ? 8 : 16;
    lds += (CHUNK / 64) * 8 + CHUNK * 2 + (size_t)ns_t * (na + 1) * 8 + 16;
    if (ns <= 4) { if (nc4) KG(4, 4); else KG(4, 8); }
    else if (ns <= 8) { if (nc4) KG(8, 4); else KG(8, 8); }
    else if (ns <= 16) { if (nc4) KG(16, 4); else KG(16, 8); }
    else return (int)hipErrorInvalidValue;
  }
#undef KL
#undef KG
  hipLaunchKernelGGL(k_reduce, dim3(nv), dim3(WG), 0, s,
                     dev_scratch, grid, nv, dev_out, naggs1, out_stride,
                     dev_plan);
  err = hipGetLastError();
  return (int)err;
}
