/*
 * jit.cpp — query-compiled scan kernels via hipRTC.
 *
 * The MI355X-native analogue of the reference's WholeStageCodegen
 * (SnappySession.scala:2388-2460 compiles one Java class per query stage;
 * here one gfx950 kernel per plan signature).  Motivation is measured, not
 * aesthetic: a probe kernel with Q1's exact shape and the plan constants
 * compile-time reaches 5.7 TB/s, while the best runtime-plan kernel stops at
 * 1.76 TB/s — literal predicates/aggregates let the compiler
 * CSE shared factors, keep every parameter in registers and emit zero
 * uniform branches in the row loop.
 *
 * Scope: plans whose unskipped batches are all CLEAN (no nulls, deletes or
 * patches) with uniform stageable column kinds, <= 4 double predicates, no
 * join, dictionary-slot or keyless grouping.  Everything else falls back to
 * the interpreted kernels (still GPU — never CPU).
 *
 * Compiled modules cache per engine by plan SHAPE: predicate bounds and
 * aggregate coefficients are tokenized (read from the cached device plan
 * at run time), so different literal values of the same shape reuse one
 * compiled kernel — the reference's ParamLiteral tokenization.
 */
#include <hip/hip_runtime.h>
#include <hip/hiprtc.h>

#include <cstdarg>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <map>
#include <mutex>
#include <string>
#include <vector>

#include "engine_internal.h"

struct JitFn {
  hipModule_t mod = nullptr;
  hipFunction_t fn = nullptr;
};

struct JitCache {
  std::map<uint64_t, JitFn> fns;
  std::mutex mu;
};

extern "C" void *sn_jit_cache_create(void) { return new JitCache(); }
extern "C" void sn_jit_cache_destroy(void *c) {
  auto *jc = (JitCache *)c;
  if (!jc) return;
  for (auto &kv : jc->fns)
    if (kv.second.mod) (void)hipModuleUnload(kv.second.mod);
  delete jc;
}

static void emitf(std::string &o, const char *fmt, ...) {
  char buf[1024];
  va_list ap; va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  o += buf;
}

/* Generate the specialized kernel source.
 * kinds[c]: SN_K_* per used column slot (uniform across batches).
 * Layout contract identical to the interpreted kernels:
 *   keyless: scratch row [2*na_t+1] = [sums][counts][rowcount]
 *   grouped: scratch row [nslots*(naggs+1)], rowcount at +naggs.         */
static std::string gen_source(const sn_dev_plan *p, const int *kinds,
                              int nslots, int na_t, int has_del) {
  const int NC = p->nused;
  const int NA = p->naggs;
  /* sparse: open-address hash aggregate — accumulator IS `out` (hacc),
   * key table and flag words arrive via the jkeys/jpayload params, and the
   * capacity is tokenized (read from the device plan) so grow-and-retry
   * reuses one compiled kernel */
  const int sparse_mode = p->sparse != 0;
  /* radix two-pass: instead of probing the global table per row (64 B
   * random lines across tens of MB — the measured 182 B/row), pass 1
   * scatters (key, agg values) records into 1 << (hcap_log2-12) hash
   * partitions via an LDS-histogram multi-split; k_radix_agg then
   * aggregates each partition into its private L2-resident table segment */
  const int radix_mode = sparse_mode && p->radix != 0;
  /* pac: per-agg-count accumulator rows ([sums][counts][rowcount]).  In
   * the JIT this occurs only for MIN/MAX plans (null-carrying batches are
   * never JIT-eligible, so counts == rowcount and the count cells simply
   * mirror the rowcount); mm plans route through the grouped layouts,
   * keyless included (one slot). */
  const int pac_mode = p->pac != 0;
  int mm_any = 0;
  for (int a = 0; a < NA; a++) mm_any |= p->aggs[a].op != 0;
  const int NS = nslots < 1 ? 1 : nslots;
  const int NA1 = pac_mode ? 2 * NA + 1 : NA + 1;
  const int grouped = nslots > 1 || sparse_mode || pac_mode;
  std::string o;
  o += R"(
typedef double double2_t __attribute__((ext_vector_type(2)));
typedef int int2_t __attribute__((ext_vector_type(2)));
typedef float float2_t __attribute__((ext_vector_type(2)));
#define GAS __attribute__((address_space(1)))
typedef unsigned long long u64;
typedef long long i64;
/* layout mirror of engine_internal.h (pointers + i32 fields only) */
struct sn_dev_col {
  const void *body; const u64 *nullw; const unsigned *nullpfx;
  const int *dictmap; const u64 *patch_bm; const int *patch_pos;
  const double *patch_val; const u64 *patch_nullbm;
  const int *rle_ends; const double *rle_vals; int rle_n;
  int patch_n; int kind; int has_nulls; int null_gid;
};
struct sn_dev_batch { int num_rows; int clean; const u64 *del_bm; sn_dev_col cols[)";
  emitf(o, "%d", SN_DEV_MAX_COLS);
  o += R"(]; };
struct sn_dev_tile { int batch; int row_start; };
struct sn_dev_pred_d { double lo, hi; int cslot, _p; };
struct sn_dev_pred_i { i64 lo, hi; int cslot, _p; };
struct sn_dev_agg { double a0, m0, a1, m1, a2, m2; int c0, c1, c2, nf, op, _p2; };
struct sn_dev_plan {
  int npreds_d, npreds_i, naggs, ngroup, nslots, nused;
  unsigned i64_mask; int gcol[2];
  const i64 *jkeys; const int *jpayload;
  int jcap_log2, jcslot, jmode, jslot_mul;
  const int *jlut; i64 jlut_min, jlut_max;
  i64 gbase[2]; int gmul0, _pad2;
  sn_dev_pred_d preds_d[8];
  sn_dev_pred_i preds_i[4];
  sn_dev_agg aggs[12];
  struct { const u64 *bm; const i64 *list; i64 base;
           int nwords, n, cslot, _p; } inp[2];
  int npreds_in, _pad4b;
  i64 *hkeys; double *hacc; int *hflags;
  int hcap_log2, sparse, pac, _pad3;
  double *precs; int *pcount; int percap, radix;
};
__device__ __forceinline__ double wsum(double x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, 64);
  return x;
}
__device__ __forceinline__ u64 f64ord(double x) {
  u64 b = (u64)__double_as_longlong(x);
  return (b & 0x8000000000000000ull) ? ~b : (b | 0x8000000000000000ull);
}
__device__ __forceinline__ double wmin(double x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmin(x, __shfl_down(x, off, 64));
  return x;
}
__device__ __forceinline__ double wmax(double x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmax(x, __shfl_down(x, off, 64));
  return x;
}
__device__ __forceinline__ u64 mix64(u64 x) {
  x += 0x9E3779B97f4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
)";
  emitf(o, "#define CHUNK %d\n#define WG %d\n#define TILE %d\n",
        1024, 256, SN_TILE_ROWS);
  /* >8 slots: block-level LDS accumulators (f64 LDS atomics) instead of
   * per-lane registers — the high-cardinality SHAMap analogue.  Occupancy 1
   * (the LDS image + accumulator array leave no room for a second group). */
  const int glob_mode = (grouped && nslots > 1024) || sparse_mode;
  const int lds_mode = grouped && nslots > 8 && !glob_mode;
  /* wbin eligibility: NA<=2 only.  Widening to Q1's 6-slot/5-agg shape was
   * MEASURED SLOWER (3.51 vs 5.13 TB/s on Q1 SF10): 6 LDS f64 atomics/row
   * across 16-lane bins conflict harder than the 36 slot-predicated fmas
   * cost, so the register accumulators stay the measured optimum for
   * multi-aggregate small-slot shapes. */
  const int wbin_pre = grouped && !lds_mode && !glob_mode && NA <= 2 &&
                       !mm_any && nslots * (NA + 1) >= 12;
  /* predicate folding: evaluate every predicate AT STAGE TIME from the
   * just-loaded registers and ballot the verdicts into a 16-word LDS
   * bitmap; the row pass then reads ONE broadcast bit per row instead of
   * re-reading each predicate column from LDS and comparing.  BUILT AND
   * MEASURED SLOWER (same-box A/B, 2 rounds each): Q6 SF100 74.9 vs
   * 75.8%, Q1 SF100 71.9 vs 73.6% — the saved LDS reads/compares were
   * already hidden under HBM latency, while the stage-time ballots add
   * v_cmp+SGPR traffic on the critical staging path.  Kept behind
   * SN_JIT_FOLD=1; default off. */
  static const int fold_env = [] {
    const char *v = getenv("SN_JIT_FOLD");
    return v && v[0] == '1';
  }();
  const int fold_preds = fold_env && !sparse_mode &&
                         (p->npreds_d + p->npreds_i) > 0;
  emitf(o, "extern \"C\" __global__ __launch_bounds__(WG, %d)\n"
           "void jit_scan(const sn_dev_batch *__restrict__ batches,\n"
           "              const sn_dev_tile *__restrict__ tiles, int ntiles,\n"
           "              double *__restrict__ out,\n"
           "              const i64 *__restrict__ jkeys_p,\n"
           "              const int *__restrict__ jpayload_p,\n"
           "              const int *__restrict__ jlut_p,\n"
           "              const sn_dev_plan *__restrict__ plan_p) {\n"
           "  const GAS i64 *jkeys = (const GAS i64 *)(u64)jkeys_p;\n"
           "  const GAS int *jpayload = (const GAS int *)(u64)jpayload_p;\n"
           "  const GAS int *jlut = (const GAS int *)(u64)jlut_p;\n"
           "  const GAS sn_dev_plan *P = (const GAS sn_dev_plan *)(u64)plan_p;\n"
           "  (void)jkeys; (void)jpayload; (void)jlut; (void)P;\n",
        /* sparse is random-probe latency-bound: more waves cover the
         * dependent loads (its LDS footprint is only the sval image).
         * SN_JIT_SPOCC overrides for occupancy sweeps. */
        [&] {
          static const int so = [] {
            const char *v = getenv("SN_JIT_SPOCC");
            return v ? atoi(v) : 0;
          }();
          if (sparse_mode) return so > 0 ? so : 4;
          return (lds_mode || glob_mode) ? 1 : (wbin_pre ? 4 : 2);
        }());
  /* tokenized plan values (the reference's ParamLiteral tokenization,
   * TokenizationTest / SnappySession plan cache): predicate bounds and
   * aggregate coefficients load once per wave from the cached device plan
   * (uniform scalar loads hoisted to kernel entry), so one compiled
   * kernel serves every literal value of the same plan SHAPE.  Structure
   * (counts, columns, neutral factors, group/join/LUT geometry) stays
   * compile-time. */
  for (int i = 0; i < p->npreds_d; i++)
    emitf(o, "  const double pd%d_lo = P->preds_d[%d].lo, pd%d_hi = P->preds_d[%d].hi;\n",
          i, i, i, i);
  for (int i = 0; i < p->npreds_i; i++)
    emitf(o, "  const i64 pi%d_lo = P->preds_i[%d].lo, pi%d_hi = P->preds_i[%d].hi;\n",
          i, i, i, i);
  for (int i = 0; i < p->npreds_in && i < 2; i++)
    emitf(o, "  const GAS u64 *inbm%d = (const GAS u64 *)(u64)P->inp[%d].bm;\n"
             "  const i64 inb%d_base = P->inp[%d].base;\n"
             "  const i64 inb%d_lim = (i64)P->inp[%d].nwords * 64;\n",
          i, i, i, i, i, i);
  /* trivial factors (a=0, m=1 — plain column reads, the common case) fold
   * back to the raw product; their triviality is part of the shape hash */
  auto ftriv = [](double a, double m) { return a == 0.0 && m == 1.0; };
  for (int a = 0; a < NA; a++) {
    const sn_dev_agg &A = p->aggs[a];
    if (A.nf >= 1 && !ftriv(A.a0, A.m0))
      emitf(o, "  const double ag%d_a0 = P->aggs[%d].a0, ag%d_m0 = P->aggs[%d].m0;\n", a, a, a, a);
    if (A.nf >= 2 && !ftriv(A.a1, A.m1))
      emitf(o, "  const double ag%d_a1 = P->aggs[%d].a1, ag%d_m1 = P->aggs[%d].m1;\n", a, a, a, a);
    if (A.nf >= 3 && !ftriv(A.a2, A.m2))
      emitf(o, "  const double ag%d_a2 = P->aggs[%d].a2, ag%d_m2 = P->aggs[%d].m2;\n", a, a, a, a);
  }
  /* wbin mode: few aggregates over many slots makes the per-slot
   * select-accumulate VALU-bound (star join: 8 slots x 2 updates vs 2 LDS
   * atomics per row) — per-wave LDS bins shift the work to the LDS pipe,
   * which the row phase barely uses. */
  const int wbin_mode = wbin_pre;
  /* fused register pass (the star-join shape): predicate-free dense-LUT
   * group-by-attr joins skip the LDS image ENTIRELY — probe, mask and
   * accumulate run straight from the staged registers with no barriers in
   * the chunk loop.  BUILT AND MEASURED SLOWER than the staged-LDS scheme
   * on star-join SF10 (2.39 TB/s staged-LDS vs 1.84 single-buffered /
   * 1.72 value-pipelined / 1.76 probe-pipelined fused): without the LDS
   * stage the per-chunk dependency chain load->probe->atomic exposes
   * whichever leg is not double-buffered, and the block-wide barrier the
   * fusion removes was evidently not the bottleneck.  Kept behind
   * SN_JIT_FUSE=1 for re-evaluation; default off. */
  static const int fuse_env = [] {
    const char *v = getenv("SN_JIT_FUSE");
    return v && v[0] == '1';
  }();
  const int fuse_mode = fuse_env && wbin_mode && p->jkeys && p->jlut &&
                        p->jmode == 1 && p->jslot_mul == 0 &&
                        p->npreds_d + p->npreds_i == 0;
  /* LDS-packed dense LUT: the star probe's dependent gathers stall on
   * L2/HBM latency (measured 58% SQ_WAIT_INST issue-stall).  When the
   * payload fits 4 bits (group-by-attr gid <= 14, or semi-join presence)
   * and the packed span fits LDS, each workgroup packs the int32 LUT
   * into nibbles ONCE at kernel start and probes LDS instead.
   * SN_JIT_SLUT=0 reverts to the register-staged L2 probe. */
  static const int slut_env = [] {
    const char *v = getenv("SN_JIT_SLUT");
    return !v || v[0] != '0';
  }();
  const long long slut_span = (p->jkeys && p->jlut)
      ? (long long)(p->jlut_max - p->jlut_min + 1) : 0;
  const long long slut_words = (slut_span + 7) / 8;
  const int slut_mode = slut_env && !fuse_mode && slut_span > 0 &&
                        (p->jmode == 0 || nslots <= 14) &&
                        slut_words * 4 <= 100 * 1024;
  if (!fuse_mode) {
    emitf(o, "  __shared__ __attribute__((aligned(16))) double sval[%d][CHUNK];\n", NC);
    if (fold_preds)
      o += "  __shared__ unsigned long long sbits[16];\n";
  }
  if (radix_mode)
    o += "  __shared__ int phist[4096];\n";   /* npart <= 4096 (cap 2^24) */
  if (!lds_mode && !wbin_mode && !glob_mode)
    emitf(o, "  __shared__ __attribute__((aligned(16))) double bacc[%d];\n",
          grouped ? NS * NA1 : 2 * na_t + 1);
  o += "  const int tid = threadIdx.x;\n";

  /* accumulators */
  if (sparse_mode) {
    /* single HBM accumulator (sparse keys -> low per-address contention;
     * probe position indexes the rows); capacity tokenized from the plan */
    emitf(o, "  GAS double *gacc = (GAS double *)(u64)out;\n"
             "  GAS i64 *hkeys = (GAS i64 *)(u64)jkeys_p;\n"
             "  GAS int *hflags = (GAS int *)(u64)jpayload_p;\n"
             "  const int hcl = P->hcap_log2;\n"
             "  const unsigned hmask = (1u << hcl) - 1;\n"
             "  const int hcap = 1 << hcl;\n"
             "  (void)gacc; (void)hkeys; (void)hmask; (void)hcap;\n");
    if (radix_mode)
      o += "  GAS double *precs = (GAS double *)(u64)P->precs;\n"
           "  GAS int *pcount = (GAS int *)(u64)P->pcount;\n"
           "  const int percap = P->percap;\n"
           "  const int npart = 1 << (hcl - P->radix);\n";
  } else if (glob_mode) {
    /* HBM accumulator IS the out/scratch pointer (host-zeroed), privatized
     * 8 ways by XCD (blockIdx & 7 matches the dispatch round-robin) so
     * same-slot atomics from different XCDs never contend and stay in the
     * local L2; k_reduce folds the 8 copies */
    emitf(o, "  GAS double *gacc = (GAS double *)(u64)out"
             " + (u64)(blockIdx.x & 7) * %d;\n", NS * NA1);
  } else if (lds_mode) {
    /* block-level LDS accumulator, initialized once (min/max cells take
     * their ord-encoding identities), flushed once at the end */
    emitf(o, "  __shared__ __attribute__((aligned(16))) double gacc[%d];\n",
          NS * NA1);
    if (!mm_any) {
      emitf(o, "  for (int i = tid; i < %d; i += WG) gacc[i] = 0.0;\n",
            NS * NA1);
    } else {
      emitf(o, "  for (int i = tid; i < %d; i += WG) {\n"
               "    const int a = i %% %d;\n"
               "    double iv = 0.0;\n", NS * NA1, NA1);
      for (int a = 0; a < NA; a++)
        if (p->aggs[a].op != 0)
          emitf(o, "    if (a == %d) iv = __longlong_as_double(%s);\n", a,
                p->aggs[a].op == 1 ? "0xFFF8000000000000ll"
                                   : "0x0000000000000000ll");
      o += "    gacc[i] = iv;\n  }\n";
    }
    o += "  __syncthreads();\n";
  } else if (wbin_mode) {
    /* 16 bins of 16 lanes each: ~2-way LDS atomic conflicts instead of the
     * 8-way a per-wave bin sees (measured: issue-stall bound, 58% of wave
     * cycles in SQ_WAIT_INST_ANY with per-wave bins) */
    emitf(o, "  __shared__ __attribute__((aligned(16))) double wbin[16][%d];\n"
             "  for (int i = tid; i < 16 * %d; i += WG)\n"
             "    ((double *)wbin)[i] = 0.0;\n"
             "  __syncthreads();\n",
          NS * NA1, NS * NA1);
  } else if (grouped) {
    emitf(o, "  double sums[%d][%d]; double rc[%d];\n", NS, NA, NS);
    emitf(o, "#pragma unroll\n  for (int s = 0; s < %d; s++) {\n"
             "    rc[s] = 0;\n", NS);
    for (int a = 0; a < NA; a++) {
      const int op = p->aggs[a].op;
      if (op == 0) emitf(o, "    sums[s][%d] = 0.0;\n", a);
      else emitf(o, "    sums[s][%d] = __longlong_as_double(%s);\n", a,
                 op == 1 ? "0x7FF0000000000000ll" : "0xFFF0000000000000ll");
    }
    o += "  }\n";
  } else {
    emitf(o, "  double sums[%d], cnts[%d], rcnt = 0.0;\n", NA, NA);
    emitf(o, "#pragma unroll\n  for (int a = 0; a < %d; a++) { sums[a] = 0; cnts[a] = 0; }\n", NA);
  }

  if (p->jkeys && p->jlut && !fuse_mode && !slut_mode)
    o += "  __shared__ int spay[CHUNK];\n";
  if (slut_mode)
    emitf(o, "  __shared__ unsigned slut[%lld];\n", slut_words);

  /* staged register buffers: per column, by width class (fuse mode adds a
   * second set, B, so the previous chunk processes under the current
   * chunk's in-flight loads) */
  for (int c = 0; c < NC; c++) {
    int k = kinds[c];
    const char *ty = (k == SN_K_F64 || k == SN_K_I64) ? "double2_t"
                     : (k == SN_K_I32 || k == SN_K_F32 || k == SN_K_DICT32)
                           ? "int2_t" : "unsigned";
    emitf(o, "  %s st%d_0, st%d_1;\n", ty, c, c);
    if (fuse_mode) emitf(o, "  %s stB%d_0, stB%d_1;\n", ty, c, c);
  }

  /* stage_load body, emitted in the prologue and inside the chunk loop;
   * `pre` picks the destination register set ("st" or fuse-mode "stB") */
  auto emit_load_pre = [&](const char *base_expr, const char *ind,
                           const char *pre) {
    for (int c = 0; c < NC; c++) {
      int k = kinds[c];
      if (k == SN_K_F64 || k == SN_K_I64) {
        emitf(o, "%s%s%d_0 = ((const GAS double2_t *)(body%d + (u64)(%s) * 8))[tid];\n", ind, pre, c, c, base_expr);
        emitf(o, "%s%s%d_1 = ((const GAS double2_t *)(body%d + (u64)(%s) * 8))[tid + WG];\n", ind, pre, c, c, base_expr);
      } else if (k == SN_K_I32 || k == SN_K_F32 || k == SN_K_DICT32) {
        emitf(o, "%s%s%d_0 = ((const GAS int2_t *)(body%d + (u64)(%s) * 4))[tid];\n", ind, pre, c, c, base_expr);
        emitf(o, "%s%s%d_1 = ((const GAS int2_t *)(body%d + (u64)(%s) * 4))[tid + WG];\n", ind, pre, c, c, base_expr);
      } else {
        emitf(o, "%s%s%d_0 = ((const GAS unsigned *)(body%d + (u64)(%s) * 2))[tid];\n", ind, pre, c, c, base_expr);
        emitf(o, "%s%s%d_1 = ((const GAS unsigned *)(body%d + (u64)(%s) * 2))[tid + WG];\n", ind, pre, c, c, base_expr);
      }
    }
  };
  auto emit_load = [&](const char *base_expr, const char *ind) {
    emit_load_pre(base_expr, ind, "st");
  };

  /* value of column c at staged-register row j (j: 0 -> st_0.x row
   * base+2*tid, 1 -> st_0.y, 2 -> st_1.x row base+2*(tid+WG), 3 -> .y) */
  auto rexpr = [&](int c, int j) -> std::string {
      char buf[96];
      const int h = j >> 1, lo = !(j & 1);
      const char *xy = (j & 1) ? "y" : "x";
      switch (kinds[c]) {
        case SN_K_F64:
          snprintf(buf, 96, "st%d_%d.%s", c, h, xy); break;
        case SN_K_I64:
          snprintf(buf, 96, "(double)__double_as_longlong(st%d_%d.%s)", c, h, xy); break;
        case SN_K_I32:
          snprintf(buf, 96, "(double)st%d_%d.%s", c, h, xy); break;
        case SN_K_F32:
          snprintf(buf, 96, "(double)((const float *)&st%d_%d)[%d]", c, h, j & 1); break;
        case SN_K_DICT32:
          snprintf(buf, 96, "(double)dm%d[st%d_%d.%s]", c, c, h, xy); break;
        case SN_K_DICT16:
          snprintf(buf, 96, lo ? "(double)dm%d[st%d_%d & 0xffff]"
                               : "(double)dm%d[st%d_%d >> 16]", c, c, h); break;
        default: /* I16 */
          snprintf(buf, 96, lo ? "(double)(short)(st%d_%d & 0xffff)"
                               : "(double)(short)(st%d_%d >> 16)", c, h); break;
      }
      return buf;
    };
    auto kexpr_reg = [&](int c, int j) -> std::string {
      char buf[96];
      const int h = j >> 1, lo = !(j & 1);
      const char *xy = (j & 1) ? "y" : "x";
      const int is_i64 = (p->i64_mask >> c) & 1u;
      switch (kinds[c]) {
        case SN_K_I32: snprintf(buf, 96, "(i64)st%d_%d.%s", c, h, xy); break;
        case SN_K_F64: case SN_K_I64:
          snprintf(buf, 96, "%s(st%d_%d.%s)",
                   is_i64 || kinds[c] == SN_K_I64 ? "__double_as_longlong" : "(i64)",
                   c, h, xy);
          break;
        default:
          snprintf(buf, 96, lo ? "(i64)(short)(st%d_%d & 0xffff)"
                               : "(i64)(short)(st%d_%d >> 16)", c, h); break;
      }
      return buf;
  };

  if (fuse_mode) {
    auto va_fused = [&](int a, const std::string (&f)[3]) -> std::string {
      const sn_dev_agg &A = p->aggs[a];
      if (A.nf < 1) return "1.0";
      char t0[192], t1[192], t2[192];
      if (ftriv(A.a0, A.m0)) snprintf(t0, 192, "%s", f[0].c_str());
      else snprintf(t0, 192, "__builtin_fma(ag%d_m0, %s, ag%d_a0)", a, f[0].c_str(), a);
      std::string r = t0;
      if (A.nf >= 2) {
        if (ftriv(A.a1, A.m1)) snprintf(t1, 192, "%s", f[1].c_str());
        else snprintf(t1, 192, "__builtin_fma(ag%d_m1, %s, ag%d_a1)", a, f[1].c_str(), a);
        r += " * "; r += t1;
      }
      if (A.nf >= 3) {
        if (ftriv(A.a2, A.m2)) snprintf(t2, 192, "%s", f[2].c_str());
        else snprintf(t2, 192, "__builtin_fma(ag%d_m2, %s, ag%d_a2)", a, f[2].c_str(), a);
        r += " * "; r += t2;
      }
      return r;
    };
    /* one row's probe + accumulate */
    auto emit_row = [&](const std::string &key, const char *rowi,
                        const std::string (&vas)[12]) {
      emitf(o, "      { const i64 fk = %s;\n"
               "        const int inr = (fk >= %lldll) & (fk <= %lldll);\n"
               "        const i64 ck = fk < %lldll ? %lldll : (fk > %lldll ? %lldll : fk);\n"
               "        int pay = inr ? jlut[ck - %lldll] : -1;\n"
               "        int okj = pay >= 0;\n",
            key.c_str(),
            (long long)p->jlut_min, (long long)p->jlut_max,
            (long long)p->jlut_min, (long long)p->jlut_min,
            (long long)p->jlut_max, (long long)p->jlut_max,
            (long long)p->jlut_min);
      if (has_del)
        emitf(o, "        if (del) { const int gr = %s;\n"
                 "          okj &= (int)(~(del[(u64)gr >> 6] >> (gr & 63)) & 1ull); }\n",
              rowi);
      emitf(o, "        if (okj) {\n"
               "          double *rw = &wbin[tid >> 4][(pay > 0 ? pay : 0) * %d];\n"
               "          atomicAdd(&rw[%d], 1.0);\n", NA + 1, NA);
      for (int a = 0; a < NA; a++)
        emitf(o, "          atomicAdd(&rw[%d], %s);\n", a, vas[a].c_str());
      o += "        }\n      }\n";
    };

    o += R"(
  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + TILE, num_rows);
)";
    if (has_del)
      o += "    const GAS u64 *del = (const GAS u64 *)(u64)b.del_bm;\n";
    for (int c = 0; c < NC; c++) {
      emitf(o, "    const GAS char *body%d = (const GAS char *)(unsigned long long)b.cols[%d].body;\n", c, c);
      if (kinds[c] == SN_K_DICT16 || kinds[c] == SN_K_DICT32)
        emitf(o, "    const GAS int *dm%d = (const GAS int *)(unsigned long long)b.cols[%d].dictmap;\n", c, c);
    }
    /* probe block: LUT gathers issued from the B set's key registers RIGHT
     * AFTER the loads, so they fly a full pipeline stage before the
     * accumulate consumes them (the round-1 measured optimum, minus LDS) */
    std::string probe;
    {
      std::string keep = std::move(o);
      o.clear();
      for (int j = 0; j < 4; j++) {
        /* kexpr against the B set: build from kexpr_reg then prefix-swap is
         * fragile; emit directly with the helper on a temp name */
        std::string kx = kexpr_reg(p->jcslot, j);
        /* retarget st{c}_ -> stB{c}_ (the key expr references exactly one
         * register name, emitted by kexpr_reg as "st<digit>") */
        size_t ppos = 0;
        while ((ppos = kx.find("st", ppos)) != std::string::npos) {
          if (ppos + 2 < kx.size() && kx[ppos + 2] >= '0' && kx[ppos + 2] <= '9')
            kx.insert(ppos + 2, "B");
          ppos += 3;
        }
        char pd[32];
        snprintf(pd, 32, "payB%d", j);
        emitf(o, "        { const i64 fk = %s;\n"
                 "          const int inr = (fk >= %lldll) & (fk <= %lldll);\n"
                 "          const i64 ck = fk < %lldll ? %lldll : (fk > %lldll ? %lldll : fk);\n"
                 "          %s = inr ? jlut[ck - %lldll] : -1; }\n",
              kx.c_str(),
              (long long)p->jlut_min, (long long)p->jlut_max,
              (long long)p->jlut_min, (long long)p->jlut_min,
              (long long)p->jlut_max, (long long)p->jlut_max,
              pd, (long long)p->jlut_min);
      }
      probe = std::move(o);
      o = std::move(keep);
    }
    /* accumulate block: consumes the A set + carried payA payloads */
    std::string proc;
    {
      std::string keep = std::move(o);
      o.clear();
      for (int j = 0; j < 4; j++) {
        char rowi[48];
        if (j < 2) snprintf(rowi, 48, "pbase + 2 * tid + %d", j);
        else snprintf(rowi, 48, "pbase + 2 * (tid + WG) + %d", j & 1);
        std::string vas[12];
        for (int a = 0; a < NA; a++) {
          const sn_dev_agg &A = p->aggs[a];
          std::string f3[3];
          f3[0] = A.nf >= 1 ? rexpr(A.c0, j) : "1.0";
          f3[1] = A.nf >= 2 ? rexpr(A.c1, j) : "1.0";
          f3[2] = A.nf >= 3 ? rexpr(A.c2, j) : "1.0";
          vas[a] = va_fused(a, f3);
        }
        emitf(o, "      { const int pay = payA%d;\n"
                 "        int okj = pay >= 0;\n", j);
        if (has_del)
          emitf(o, "        if (del) { const int gr = %s;\n"
                   "          okj &= (int)(~(del[(u64)gr >> 6] >> (gr & 63)) & 1ull); }\n",
                rowi);
        emitf(o, "        if (okj) {\n"
                 "          double *rw = &wbin[tid >> 4][(pay > 0 ? pay : 0) * %d];\n"
                 "          atomicAdd(&rw[%d], 1.0);\n", NA + 1, NA);
        for (int a = 0; a < NA; a++)
          emitf(o, "          atomicAdd(&rw[%d], %s);\n", a, vas[a].c_str());
        o += "        }\n      }\n";
      }
      proc = std::move(o);
      o = std::move(keep);
    }
    std::string bcopy;
    for (int c = 0; c < NC; c++) {
      char cb[96];
      snprintf(cb, 96, "        st%d_0 = stB%d_0; st%d_1 = stB%d_1;\n",
               c, c, c, c);
      bcopy += cb;
    }
    bcopy += "        payA0 = payB0; payA1 = payB1;"
             " payA2 = payB2; payA3 = payB3;\n";
    o += "  int payA0 = -1, payA1 = -1, payA2 = -1, payA3 = -1;\n"
         "  int payB0, payB1, payB2, payB3;\n";
    o += "    int pbase = -1;\n"
         "    for (int base = tile.row_start; base < tile_end; base += CHUNK) {\n"
         "      const int rows = min(CHUNK, tile_end - base);\n"
         "      if (rows == CHUNK) {\n";
    /* issue this chunk's loads into B, accumulate the PREVIOUS chunk from
     * A (probes already in flight since last iteration), then probe B's
     * keys (gathers fly through the next load+accumulate) and rotate */
    emit_load_pre("base", "        ", "stB");
    o += "        if (pbase >= 0) {\n" + proc + "        }\n" + probe + bcopy +
         "        pbase = base;\n";
    /* scalar tail: drain the pipeline, then direct global reads */
    o += "      } else {\n"
         "        if (pbase >= 0) {\n" + proc + "          pbase = -1;\n"
         "        }\n"
         "        for (int r = tid; r < rows; r += WG) {\n"
         "          const int gr = base + r;\n";
    auto sexpr = [&](int c) -> std::string {
      char buf[96];
      switch (kinds[c]) {
        case SN_K_F64: snprintf(buf, 96, "((const GAS double *)body%d)[gr]", c); break;
        case SN_K_I64: snprintf(buf, 96, "(double)((const GAS i64 *)body%d)[gr]", c); break;
        case SN_K_I32: snprintf(buf, 96, "(double)((const GAS int *)body%d)[gr]", c); break;
        case SN_K_F32: snprintf(buf, 96, "(double)((const GAS float *)body%d)[gr]", c); break;
        case SN_K_DICT32: snprintf(buf, 96, "(double)dm%d[((const GAS int *)body%d)[gr]]", c, c); break;
        case SN_K_DICT16: snprintf(buf, 96, "(double)dm%d[(int)((const GAS unsigned short *)body%d)[gr]]", c, c); break;
        default: snprintf(buf, 96, "(double)((const GAS short *)body%d)[gr]", c); break;
      }
      return buf;
    };
    {
      char kb[96];
      const int jc = p->jcslot;
      const int is_i64 = (p->i64_mask >> jc) & 1u;
      switch (kinds[jc]) {
        case SN_K_I32: snprintf(kb, 96, "(i64)((const GAS int *)body%d)[gr]", jc); break;
        case SN_K_F64: case SN_K_I64:
          snprintf(kb, 96, "%s((const GAS double *)body%d)[gr]%s",
                   is_i64 || kinds[jc] == SN_K_I64 ? "__double_as_longlong(" : "(i64)(",
                   jc, ")");
          break;
        default: snprintf(kb, 96, "(i64)((const GAS short *)body%d)[gr]", jc); break;
      }
      std::string vas[12];
      for (int a = 0; a < NA; a++) {
        const sn_dev_agg &A = p->aggs[a];
        std::string f3[3];
        f3[0] = A.nf >= 1 ? sexpr(A.c0) : "1.0";
        f3[1] = A.nf >= 2 ? sexpr(A.c1) : "1.0";
        f3[2] = A.nf >= 3 ? sexpr(A.c2) : "1.0";
        vas[a] = va_fused(a, f3);
      }
      emit_row(kb, "gr", vas);
    }
    o += "        }\n"
         "      }\n"
         "    }\n"
         "    if (pbase >= 0) {\n" + proc + "    }\n"
         "  }\n";
    int nvf = nslots * (NA + 1);
    emitf(o, "  __syncthreads();\n"
             "  for (int i = tid; i < %d; i += WG) {\n"
             "    double acc = 0.0;\n"
             "#pragma unroll\n"
             "    for (int w = 0; w < 16; w++) acc += wbin[w][i];\n"
             "    out[(u64)blockIdx.x * %d + i] = acc;\n"
             "  }\n"
             "}\n", nvf, nvf);
    return o;
  }

  if (slut_mode) {
    /* pack the dense LUT into 4-bit LDS nibbles, once per workgroup:
     * 15 = absent; group-by-attr payloads are gids <= 14 (structural
     * gate), semi-join presence packs as 0 */
    emitf(o, "  for (unsigned i = tid; i < %lldu; i += WG) {\n"
             "    unsigned wd = 0u;\n"
             "#pragma unroll\n"
             "    for (int j = 0; j < 8; j++) {\n"
             "      const long long ix = (long long)i * 8 + j;\n"
             "      const int v = ix < %lldll ? jlut[ix] : -1;\n"
             "      const unsigned nib = v < 0 ? 15u : %s;\n"
             "      wd |= nib << (4 * j);\n"
             "    }\n"
             "    slut[i] = wd;\n"
             "  }\n"
             "  __syncthreads();\n",
          slut_words, slut_span, p->jmode == 0 ? "0u" : "(unsigned)v");
  }
  o += R"(
  for (int t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const sn_dev_tile tile = tiles[t];
    const sn_dev_batch &b = batches[tile.batch];
    const int num_rows = b.num_rows;
    const int tile_end = min(tile.row_start + TILE, num_rows);
)";
  /* hoist body/dict pointers */
  if (has_del)
    o += "    const GAS u64 *del = (const GAS u64 *)(u64)b.del_bm;\n";
  for (int c = 0; c < NC; c++) {
    emitf(o, "    const GAS char *body%d = (const GAS char *)(unsigned long long)b.cols[%d].body;\n", c, c);
    if (kinds[c] == SN_K_DICT16 || kinds[c] == SN_K_DICT32)
      emitf(o, "    const GAS int *dm%d = (const GAS int *)(unsigned long long)b.cols[%d].dictmap;\n", c, c);
  }

  auto emit_write = [&](const char *ind) {
    for (int c = 0; c < NC; c++) {
      int k = kinds[c];
      if (k == SN_K_F64 || k == SN_K_I64) {
        emitf(o, "%s((double2_t *)sval[%d])[tid] = st%d_0;\n", ind, c, c);
        emitf(o, "%s((double2_t *)sval[%d])[tid + WG] = st%d_1;\n", ind, c, c);
      } else if (k == SN_K_I32) {
        emitf(o, "%s{ double2_t y0; y0.x = (double)st%d_0.x; y0.y = (double)st%d_0.y;\n"
                 "%s  ((double2_t *)sval[%d])[tid] = y0;\n"
                 "%s  double2_t y1; y1.x = (double)st%d_1.x; y1.y = (double)st%d_1.y;\n"
                 "%s  ((double2_t *)sval[%d])[tid + WG] = y1; }\n",
              ind, c, c, ind, c, ind, c, c, ind, c);
      } else if (k == SN_K_F32) {
        emitf(o, "%s{ float2_t f0 = *(float2_t *)&st%d_0, f1 = *(float2_t *)&st%d_1;\n"
                 "%s  double2_t y0; y0.x = (double)f0.x; y0.y = (double)f0.y;\n"
                 "%s  double2_t y1; y1.x = (double)f1.x; y1.y = (double)f1.y;\n"
                 "%s  ((double2_t *)sval[%d])[tid] = y0;\n"
                 "%s  ((double2_t *)sval[%d])[tid + WG] = y1; }\n",
              ind, c, c, ind, ind, ind, c, ind, c);
      } else if (k == SN_K_DICT32) {
        emitf(o, "%s{ double2_t y0; y0.x = (double)dm%d[st%d_0.x]; y0.y = (double)dm%d[st%d_0.y];\n"
                 "%s  double2_t y1; y1.x = (double)dm%d[st%d_1.x]; y1.y = (double)dm%d[st%d_1.y];\n"
                 "%s  ((double2_t *)sval[%d])[tid] = y0;\n"
                 "%s  ((double2_t *)sval[%d])[tid + WG] = y1; }\n",
              ind, c, c, c, c, ind, c, c, c, c, ind, c, ind, c);
      } else if (k == SN_K_DICT16) {
        emitf(o, "%s{ double2_t y0; y0.x = (double)dm%d[st%d_0 & 0xffff]; y0.y = (double)dm%d[st%d_0 >> 16];\n"
                 "%s  double2_t y1; y1.x = (double)dm%d[st%d_1 & 0xffff]; y1.y = (double)dm%d[st%d_1 >> 16];\n"
                 "%s  ((double2_t *)sval[%d])[tid] = y0;\n"
                 "%s  ((double2_t *)sval[%d])[tid + WG] = y1; }\n",
              ind, c, c, c, c, ind, c, c, c, c, ind, c, ind, c);
      } else { /* I16 signed */
        emitf(o, "%s{ double2_t y0; y0.x = (double)(short)(st%d_0 & 0xffff); y0.y = (double)(short)(st%d_0 >> 16);\n"
                 "%s  double2_t y1; y1.x = (double)(short)(st%d_1 & 0xffff); y1.y = (double)(short)(st%d_1 >> 16);\n"
                 "%s  ((double2_t *)sval[%d])[tid] = y0;\n"
                 "%s  ((double2_t *)sval[%d])[tid + WG] = y1; }\n",
              ind, c, c, ind, c, c, ind, c, ind, c);
      }
    }
  };

  /* LUT probe straight from the staged key registers, BEFORE they are
   * written to LDS: the dependent LUT gathers then overlap the stage_write
   * + barrier + next stage_load instead of stalling the row pass
   * (measured: in-pass probing leaves 35% of wave cycles parked).
   * Register mapping: st_0 covers rows 2*tid(+1), st_1 rows 2*(tid+WG)(+1);
   * the row pass reads spay[tid + k*WG] after the barrier. */
  auto emit_lut_probe = [&](const std::string &kexpr, const std::string &dst,
                            const char *ind) {
    emitf(o, "%s{ const i64 key = %s;\n"
             "%s  const int inr = (key >= %lldll) & (key <= %lldll);\n"
             "%s  const i64 ck = key < %lldll ? %lldll : (key > %lldll ? %lldll : key);\n"
             "%s  const int pv = jlut[ck - %lldll];\n"
             "%s  %s = inr ? pv : -1; }\n",
          ind, kexpr.c_str(),
          ind, (long long)p->jlut_min, (long long)p->jlut_max,
          ind, (long long)p->jlut_min, (long long)p->jlut_min,
          (long long)p->jlut_max, (long long)p->jlut_max,
          ind, (long long)p->jlut_min,
          ind, dst.c_str());
  };
  const int use_spay = p->jkeys && p->jlut && !slut_mode;
  o += "    int staged = 0;\n"
       "    if (tile.row_start + CHUNK <= tile_end) {\n";
  emit_load("tile.row_start", "      ");
  o += "      staged = 1;\n    }\n"
       "    for (int base = tile.row_start; base < tile_end; base += CHUNK) {\n"
       "      const int rows = min(CHUNK, tile_end - base);\n";
  if (radix_mode)   /* previous chunk's scatter is behind the tail barrier */
    o += "      for (int i = tid; i < npart; i += WG) phist[i] = 0;\n";
  o += "      if (staged) {\n";
  if (use_spay) {
    const int jc = p->jcslot;
    const int jk = kinds[jc];
    int is_i64 = (p->i64_mask >> jc) & 1u;
    char b0[64], b1[64], b2[64], b3[64];
    if (jk == SN_K_I32) {
      snprintf(b0, 64, "(i64)st%d_0.x", jc); snprintf(b1, 64, "(i64)st%d_0.y", jc);
      snprintf(b2, 64, "(i64)st%d_1.x", jc); snprintf(b3, 64, "(i64)st%d_1.y", jc);
    } else if (jk == SN_K_F64 || jk == SN_K_I64) {
      const char *cv = is_i64 || jk == SN_K_I64 ? "__double_as_longlong" : "(i64)";
      snprintf(b0, 64, "%s(st%d_0.x)", cv, jc); snprintf(b1, 64, "%s(st%d_0.y)", cv, jc);
      snprintf(b2, 64, "%s(st%d_1.x)", cv, jc); snprintf(b3, 64, "%s(st%d_1.y)", cv, jc);
    } else { /* I16 */
      snprintf(b0, 64, "(i64)(short)(st%d_0 & 0xffff)", jc);
      snprintf(b1, 64, "(i64)(short)(st%d_0 >> 16)", jc);
      snprintf(b2, 64, "(i64)(short)(st%d_1 & 0xffff)", jc);
      snprintf(b3, 64, "(i64)(short)(st%d_1 >> 16)", jc);
    }
    emit_lut_probe(b0, "spay[2 * tid]", "        ");
    emit_lut_probe(b1, "spay[2 * tid + 1]", "        ");
    emit_lut_probe(b2, "spay[2 * (tid + WG)]", "        ");
    emit_lut_probe(b3, "spay[2 * (tid + WG) + 1]", "        ");
  }
  emit_write("        ");
  if (fold_preds) {
    /* evaluate every predicate from the staged registers, one wave ballot
     * per register row into the chunk bitmap */
    for (int j = 0; j < 4; j++) {
      emitf(o, "        { int okp = 1;\n");
      for (int i = 0; i < p->npreds_d; i++) {
        emitf(o, "          { const double x = %s;\n"
                 "            okp &= (x >= pd%d_lo) & (x <= pd%d_hi); }\n",
              rexpr(p->preds_d[i].cslot, j).c_str(), i, i);
      }
      for (int i = 0; i < p->npreds_i; i++) {
        emitf(o, "          { const i64 x = %s;\n"
                 "            okp &= (x >= pi%d_lo) & (x <= pi%d_hi); }\n",
              kexpr_reg(p->preds_i[i].cslot, j).c_str(), i, i);
      }
      const int w0 = (j >= 2 ? 1 : 0);   /* +4 blocks for st_1 rows */
      emitf(o, "          const u64 bl = __ballot(okp);\n"
               "          if ((tid & 63) == 0) sbits[((tid >> 6) + %d) * 2 + %d] = bl;\n"
               "        }\n", w0 * 4, j & 1);
    }
  }
  o += "      } else {\n"
       "        /* scalar tail conversion (the [rows, CHUNK) image tail is\n"
       "         * zeroed below: raw LDS can hold NaN-pattern bits from the\n"
       "         * previous kernel, and fma(0, NaN, sum) = NaN would poison\n"
       "         * the slot-predicated register accumulators) */\n";
  if (fold_preds)
    o += "        for (int i = tid; i < 16; i += WG) sbits[i] = 0ull;\n"
         "        __syncthreads();\n";
  for (int c = 0; c < NC; c++) {
    int k = kinds[c];
    emitf(o, "        for (int r = tid; r < rows; r += WG) sval[%d][r] = ", c);
    switch (k) {
      case SN_K_F64:
        emitf(o, "((const GAS double *)(body%d))[base + r];\n", c); break;
      case SN_K_I64:
        emitf(o, "((const GAS double *)(body%d))[base + r];\n", c); break; /* raw bits */
      case SN_K_I32:
        emitf(o, "(double)((const GAS int *)(body%d))[base + r];\n", c); break;
      case SN_K_F32:
        emitf(o, "(double)((const GAS float *)(body%d))[base + r];\n", c); break;
      case SN_K_DICT32:
        emitf(o, "(double)dm%d[((const GAS int *)(body%d))[base + r]];\n", c, c); break;
      case SN_K_DICT16:
        emitf(o, "(double)dm%d[(int)((const GAS unsigned short *)(body%d))[base + r]];\n", c, c); break;
      default: /* I16 */
        emitf(o, "(double)((const GAS short *)(body%d))[base + r];\n", c); break;
    }
  }
  for (int c = 0; c < NC; c++)
    emitf(o, "        for (int r = rows + tid; r < CHUNK; r += WG)"
             " sval[%d][r] = 0.0;\n", c);
  if (fold_preds) {
    /* same thread wrote these sval rows above — no barrier needed */
    o += "        for (int r = tid; r < rows; r += WG) {\n"
         "          int okp = 1;\n";
    for (int i = 0; i < p->npreds_d; i++)
      emitf(o, "          { const double x = sval[%d][r];\n"
               "            okp &= (x >= pd%d_lo) & (x <= pd%d_hi); }\n",
            p->preds_d[i].cslot, i, i);
    for (int i = 0; i < p->npreds_i; i++)
      emitf(o, "          { const i64 x = __double_as_longlong(sval[%d][r]);\n"
               "            okp &= (x >= pi%d_lo) & (x <= pi%d_hi); }\n",
            p->preds_i[i].cslot, i, i);
    o += "          if (okp)\n"
         "            atomicOr(&sbits[((r >> 7) << 1) | (r & 1)],\n"
         "                     1ull << ((r >> 1) & 63));\n"
         "        }\n";
  }
  o += "      }\n"
       "      __syncthreads();\n"
       "      const int nbase = base + CHUNK;\n"
       "      const int next_staged = nbase + CHUNK <= tile_end;\n"
       "      if (next_staged) {\n";
  emit_load("nbase", "        ");
  o += "      }\n";

  /* tail / unstaged chunks: fill spay from the LDS image (lane-local
   * mapping, so no extra barrier before the row pass reads it) */
  if (use_spay) {
    int is_i64 = (p->i64_mask >> p->jcslot) & 1u;
    emitf(o, "      if (!staged) {\n"
             "#pragma unroll\n"
             "        for (int k = 0; k < CHUNK / WG; k++) {\n"
             "          const int r = tid + k * WG;\n"
             "          const double jx = sval[%d][r];\n",
          p->jcslot);
    std::string kx = is_i64 ? "__double_as_longlong(jx)" : "(i64)jx";
    emit_lut_probe(kx, "spay[r]", "          ");
    o += "        }\n"
         "      }\n";
  }

  /* fused row pass */
  if (fold_preds)
    o += "#pragma unroll 2\n"
         "      for (int k = 0; k < CHUNK / WG; k++) {\n"
         "        const int r = tid + k * WG;\n"
         "        int ok = (int)((sbits[((r >> 7) << 1) | (r & 1)]\n"
         "                        >> ((r >> 1) & 63)) & 1ull);\n";
  else {
    if (radix_mode) {
      /* per-k register stash: pass A computes (partition, rank, key,
       * values) per row; pass B scatters after the per-chunk base
       * reservation.  Fully unrolled so the arrays stay in registers. */
      emitf(o, "      i64 rkey[CHUNK / WG];\n"
               "      int rok[CHUNK / WG], rpk[CHUNK / WG], rrnk[CHUNK / WG];\n");
      for (int a = 0; a < NA; a++)
        emitf(o, "      double rva%d[CHUNK / WG];\n", a);
      o += "#pragma unroll\n";
    } else {
      o += "#pragma unroll 2\n";
    }
    o += "      for (int k = 0; k < CHUNK / WG; k++) {\n"
         "        const int r = tid + k * WG;\n"
         "        int ok = r < rows;\n";
  }
  if (has_del)
    o += "        if (del) {\n"
         "          const int gr = base + r;\n"
         "          ok &= (int)(~(del[(u64)gr >> 6] >> (gr & 63)) & 1ull);\n"
         "        }\n";
  if (!fold_preds) {
    for (int i = 0; i < p->npreds_d; i++) {
      emitf(o, "        { const double x = sval[%d][r];\n"
               "          ok &= (x >= pd%d_lo) & (x <= pd%d_hi); }\n",
            p->preds_d[i].cslot, i, i);
    }
    for (int i = 0; i < p->npreds_i; i++) {
      emitf(o, "        { const i64 x = __double_as_longlong(sval[%d][r]);\n"
               "          ok &= (x >= pi%d_lo) & (x <= pi%d_hi); }\n",
            p->preds_i[i].cslot, i, i);
    }
  }
  for (int i = 0; i < p->npreds_in && i < 2; i++) {
    const int cs = p->inp[i].cslot;
    const int is64 = (p->i64_mask >> cs) & 1u;
    emitf(o, "        { const i64 v = %ssval[%d][r]%s;\n"
             "          const i64 ix = v - inb%d_base;\n"
             "          ok &= (ix >= 0) & (ix < inb%d_lim) &\n"
             "                (int)((inbm%d[ix >> 6] >> (ix & 63)) & 1ull); }\n",
          is64 ? "__double_as_longlong(" : "(i64)", cs, is64 ? ")" : "",
          i, i, i);
  }
  if (radix_mode)   /* before the wave skip: pass B keys off rok alone */
    o += "        rok[k] = ok;\n";
  o += "        if (__popcll(__ballot(ok)) == 0) continue;\n";
  if (p->jkeys) {
    /* broadcast-dimension probe with literal table shape (mask, key slot,
     * i64-ness); empty-slot sentinel = INT64_MIN, same as the interpreted
     * probe_sweep */
    int is_i64 = (p->i64_mask >> p->jcslot) & 1u;
    if (slut_mode) {
      /* 4-bit LDS probe: range check + one LDS read per row */
      emitf(o, "        int pay = -1;\n"
               "        {\n"
               "          const double jx = sval[%d][r];\n"
               "          const i64 key = %s;\n"
               "          const i64 ix = key - %lldll;\n"
               "          if (ix >= 0 && ix < %lldll) {\n"
               "            const unsigned nib =\n"
               "                (slut[ix >> 3] >> (((unsigned)ix & 7u) * 4u)) & 15u;\n"
               "            pay = nib == 15u ? -1 : (int)nib;\n"
               "          }\n"
               "        }\n"
               "        ok &= pay >= 0;\n",
            p->jcslot, is_i64 ? "__double_as_longlong(jx)" : "(i64)jx",
            (long long)p->jlut_min, slut_span);
    } else if (p->jlut) {
      (void)is_i64;   /* probed from the staged registers into spay */
      o += "        const int pay = spay[r];\n"
           "        ok &= pay >= 0;\n";
    } else {
      emitf(o, "        int pay = -1;\n"
               "        if (ok) {\n"
               "          const double jx = sval[%d][r];\n"
               "          const i64 key = %s;\n"
               "          unsigned h = (unsigned)mix64((u64)key) & %uu;\n"
               "          while (true) {\n"
               "            const i64 k0 = jkeys[h];\n"
               "            if (k0 == key) { pay = jpayload[h]; break; }\n"
               "            if (k0 == (i64)0x8000000000000000ll) break;\n"
               "            h = (h + 1) & %uu;\n"
               "          }\n"
               "          ok = pay >= 0;\n"
               "        }\n",
            p->jcslot,
            is_i64 ? "__double_as_longlong(jx)" : "(i64)jx",
            (1u << p->jcap_log2) - 1, (1u << p->jcap_log2) - 1);
    }
  }
  if (sparse_mode) {
    /* open-address probe (ByteBufferHashMap.putBufferIfAbsent shape):
     * key geometry is compile-time, capacity tokenized */
    if (p->ngroup == 1) {
      if ((p->i64_mask >> p->gcol[0]) & 1u)
        emitf(o, "        const i64 skey = __double_as_longlong(sval[%d][r]);\n",
              p->gcol[0]);
      else
        emitf(o, "        const i64 skey = (i64)sval[%d][r];\n", p->gcol[0]);
    } else {
      emitf(o, "        const i64 skey = (i64)(((u64)(unsigned)(int)sval[%d][r]"
               " << 32) | (unsigned)(int)sval[%d][r]);\n",
            p->gcol[0], p->gcol[1]);
    }
    if (radix_mode) {
      /* pass A: partition by mix64 HIGH bits (pass 2 probes the segment
       * with the LOW bits — independent), rank via the LDS histogram */
      o += "        rkey[k] = skey;\n"
           "        int rpk_ = 0, rrnk_ = 0;\n"
           "        if (ok) {\n"
           "          rpk_ = (int)((mix64((u64)skey) >> 44) & (u64)(npart - 1));\n"
           "          rrnk_ = atomicAdd(&phist[rpk_], 1);\n"
           "        }\n"
           "        rpk[k] = rpk_; rrnk[k] = rrnk_;\n";
    } else {
      o += R"(        int slot = -1;
        if (ok) {
          if (skey == -1ll) slot = hcap;     /* sentinel-valued key */
          else {
            unsigned h = (unsigned)mix64((u64)skey) & hmask;
            for (unsigned it = 0; it <= hmask; ++it) {
              const i64 k0 = hkeys[h];
              if (k0 == skey) { slot = (int)h; break; }
              if (k0 == -1ll) {
                const i64 old = (i64)atomicCAS((unsigned long long *)&hkeys[h],
                                               (unsigned long long)-1ll,
                                               (unsigned long long)skey);
                if (old == -1ll) { atomicAdd((int *)(hflags + 2), 1); slot = (int)h; break; }
                if (old == skey) { slot = (int)h; break; }
              }
              h = (h + 1) & hmask;
            }
            if (slot < 0) atomicOr((int *)hflags, 1);
          }
        }
        ok &= slot >= 0;
)";
    }
  } else if (grouped) {
    if (p->jkeys && p->jmode == 1) {
      if (p->jslot_mul > 0 && p->ngroup >= 1)
        /* composite GROUP BY dim_attr, fact_col (all literals) */
        emitf(o, "        const int slot = (pay > 0 ? pay : 0) * %d +"
                 " (int)((i64)sval[%d][r] - %lldll);\n",
              p->jslot_mul, p->gcol[0], (long long)p->gbase[0]);
      else
        o += "        const int slot = pay > 0 ? pay : 0;\n";
    } else if (p->ngroup == 0) {
      /* keyless plan in the grouped (pac) layout: one slot */
      o += "        const int slot = 0;\n";
    } else {
      /* slot = (v0 - base0)*mul0 + (v1 - base1), all literals; dictionary
       * keys arrive premultiplied (base 0, mul 1) so this folds to the
       * plain cast for them */
      emitf(o, "        int slot = (int)(((i64)sval[%d][r] - %lldll) * %d);\n",
            p->gcol[0], (long long)p->gbase[0], p->gmul0 ? p->gmul0 : 1);
      if (p->ngroup >= 2)
        emitf(o, "        slot += (int)((i64)sval[%d][r] - %lldll);\n",
              p->gcol[1], (long long)p->gbase[1]);
    }
  }
  /* factor read: i64 slots hold raw bits in the LDS image — emit the
   * convert only for those slots (compile-time; i64_mask is in the shape) */
  auto fread = [&](char *dst, int cs) {
    if ((p->i64_mask >> cs) & 1u)
      snprintf(dst, 72, "(double)__double_as_longlong(sval[%d][r])", cs);
    else
      snprintf(dst, 72, "sval[%d][r]", cs);
  };
  for (int a = 0; a < NA; a++) {
    const sn_dev_agg &A = p->aggs[a];
    if (A.nf < 1) { emitf(o, "        const double va%d = 1.0;\n", a); continue; }
    char t0[128], t1[128], t2[128], f0[72], f1[72], f2[72];
    fread(f0, A.c0); fread(f1, A.c1); fread(f2, A.c2);
    if (ftriv(A.a0, A.m0)) snprintf(t0, 128, "%s", f0);
    else snprintf(t0, 128, "__builtin_fma(ag%d_m0, %s, ag%d_a0)", a, f0, a);
    if (ftriv(A.a1, A.m1)) snprintf(t1, 128, "%s", f1);
    else snprintf(t1, 128, "__builtin_fma(ag%d_m1, %s, ag%d_a1)", a, f1, a);
    if (ftriv(A.a2, A.m2)) snprintf(t2, 128, "%s", f2);
    else snprintf(t2, 128, "__builtin_fma(ag%d_m2, %s, ag%d_a2)", a, f2, a);
    emitf(o, "        const double va%d = %s", a, t0);
    if (A.nf >= 2) emitf(o, " * %s", t1);
    if (A.nf >= 3) emitf(o, " * %s", t2);
    o += ";\n";
  }
  if (radix_mode) {
    for (int a = 0; a < NA; a++)
      emitf(o, "        rva%d[k] = va%d;\n", a, a);
  } else if (lds_mode || glob_mode) {
    emitf(o, "        if (ok) {\n"
             "          %sdouble *row = &gacc[(u64)slot * %d];\n"
             "          atomicAdd((double *)&row[%d], 1.0);\n",
          glob_mode ? "GAS " : "", NA1, NA1 - 1);
    for (int a = 0; a < NA; a++) {
      const int op = p->aggs[a].op;
      if (op == 0)
        emitf(o, "          atomicAdd((double *)&row[%d], va%d);\n", a, a);
      else
        emitf(o, "          %s((unsigned long long *)&row[%d], f64ord(va%d));\n",
              op == 1 ? "atomicMin" : "atomicMax", a, a);
      if (pac_mode)
        emitf(o, "          atomicAdd((double *)&row[%d], 1.0);\n", NA + a);
    }
    o += "        }\n";
  } else if (wbin_mode) {
    emitf(o, "        if (ok) {\n"
             "          double *row = &wbin[tid >> 4][slot * %d];\n"
             "          atomicAdd(&row[%d], 1.0);\n", NA + 1, NA);
    for (int a = 0; a < NA; a++)
      emitf(o, "          atomicAdd(&row[%d], va%d);\n", a, a);
    o += "        }\n";
  } else if (grouped) {
    /* m is 0.0 or 1.0; fma(m, va, sum) is bit-identical to the select+add
     * form (m=0 -> sum exactly, m=1 -> one rounding like the add) but one
     * VALU op per aggregate instead of two */
    emitf(o, "#pragma unroll\n        for (int s = 0; s < %d; s++) {\n"
             "          const double m = (ok && slot == s) ? 1.0 : 0.0;\n"
             "          rc[s] += m;\n", NS);
    for (int a = 0; a < NA; a++) {
      const int op = p->aggs[a].op;
      if (op == 0)
        emitf(o, "          sums[s][%d] = __builtin_fma(m, va%d, sums[s][%d]);\n",
              a, a, a);
      else
        emitf(o, "          sums[s][%d] = %s(sums[s][%d],\n"
                 "              (ok && slot == s) ? va%d : __longlong_as_double(%s));\n",
              a, op == 1 ? "fmin" : "fmax", a, a,
              op == 1 ? "0x7FF0000000000000ll" : "0xFFF0000000000000ll");
    }
    o += "        }\n";
  } else {
    for (int a = 0; a < NA; a++) {
      emitf(o, "        sums[%d] += ok ? va%d : 0.0;\n", a, a);
      emitf(o, "        cnts[%d] += ok ? 1.0 : 0.0;\n", a);
    }
    o += "        rcnt += ok ? 1.0 : 0.0;\n";
  }
  o += "      }\n";
  if (radix_mode) {
    /* reserve contiguous global space per (chunk, partition) — one global
     * atomic per NON-EMPTY partition per chunk, not one per row — then
     * scatter the stashed records to [base, base+count) */
    emitf(o, "      __syncthreads();\n"
             "      for (int i = tid; i < npart; i += WG) {\n"
             "        const int c = phist[i];\n"
             "        phist[i] = c ? atomicAdd(&pcount[i], c) : 0;\n"
             "      }\n"
             "      __syncthreads();\n"
             "#pragma unroll\n"
             "      for (int k = 0; k < CHUNK / WG; k++) {\n"
             "        if (!rok[k]) continue;\n"
             "        const u64 off = (u64)phist[rpk[k]] + (u64)rrnk[k];\n"
             "        if (off >= (u64)percap) { atomicOr(hflags + 3, 1); continue; }\n"
             "        GAS double *rec = precs + ((u64)rpk[k] * (u64)percap + off) * %d;\n"
             "        ((GAS i64 *)rec)[0] = rkey[k];\n", 1 + NA);
    for (int a = 0; a < NA; a++)
      emitf(o, "        rec[%d] = rva%d[k];\n", 1 + a, a);
    o += "      }\n";
  }
  o += "      __syncthreads();\n"
       "      staged = next_staged;\n"
       "    }\n"
       "  }\n";

  /* block reduce into LDS bacc then scratch row */
  int nv = grouped ? NS * NA1 : 2 * na_t + 1;
  if (glob_mode) {
    /* nothing to flush — the HBM accumulator holds the single partial set */
    o += "}\n";
    return o;
  }
  if (lds_mode) {
    /* gacc IS the block accumulator — flush it straight to the scratch row */
    emitf(o, "  __syncthreads();\n"
             "  for (int i = tid; i < %d; i += WG)\n"
             "    out[(u64)blockIdx.x * %d + i] = gacc[i];\n"
             "}\n", nv, nv);
    return o;
  }
  if (wbin_mode) {
    emitf(o, "  __syncthreads();\n"
             "  for (int i = tid; i < %d; i += WG) {\n"
             "    double acc = 0.0;\n"
             "#pragma unroll\n"
             "    for (int w = 0; w < 16; w++) acc += wbin[w][i];\n"
             "    out[(u64)blockIdx.x * %d + i] = acc;\n"
             "  }\n"
             "}\n", nv, nv);
    return o;
  }
  if (!mm_any || !grouped) {
    emitf(o, "  for (int i = tid; i < %d; i += WG) bacc[i] = 0.0;\n"
             "  __syncthreads();\n", nv);
  } else {
    emitf(o, "  for (int i = tid; i < %d; i += WG) {\n"
             "    const int a = i %% %d;\n"
             "    double iv = 0.0;\n", nv, NA1);
    for (int a = 0; a < NA; a++)
      if (p->aggs[a].op != 0)
        emitf(o, "    if (a == %d) iv = __longlong_as_double(%s);\n", a,
              p->aggs[a].op == 1 ? "0xFFF8000000000000ll"
                                 : "0x0000000000000000ll");
    o += "    bacc[i] = iv;\n  }\n  __syncthreads();\n";
  }
  if (grouped) {
    emitf(o, "#pragma unroll\n  for (int s = 0; s < %d; s++) {\n", NS);
    for (int a = 0; a < NA; a++) {
      const int op = p->aggs[a].op;
      if (op == 0)
        emitf(o, "    { double x = wsum(sums[s][%d]);\n"
                 "      if ((tid & 63) == 0 && x != 0.0) atomicAdd(&bacc[s * %d + %d], x); }\n",
              a, NA1, a);
      else
        emitf(o, "    { double x = %s(sums[s][%d]);\n"
                 "      if ((tid & 63) == 0)\n"
                 "        %s((unsigned long long *)&bacc[s * %d + %d], f64ord(x)); }\n",
              op == 1 ? "wmin" : "wmax", a,
              op == 1 ? "atomicMin" : "atomicMax", NA1, a);
    }
    emitf(o, "    { double x = wsum(rc[s]);\n"
             "      if ((tid & 63) == 0 && x != 0.0) {\n"
             "        atomicAdd(&bacc[s * %d + %d], x);\n", NA1, NA1 - 1);
    if (pac_mode)
      for (int a = 0; a < NA; a++)
        emitf(o, "        atomicAdd(&bacc[s * %d + %d], x);\n", NA1, NA + a);
    o += "      } }\n  }\n";
  } else {
    for (int a = 0; a < NA; a++) {
      emitf(o, "  { double x = wsum(sums[%d]);\n"
               "    if ((tid & 63) == 0 && x != 0.0) atomicAdd(&bacc[%d], x); }\n", a, a);
      emitf(o, "  { double x = wsum(cnts[%d]);\n"
               "    if ((tid & 63) == 0 && x != 0.0) atomicAdd(&bacc[%d], x); }\n",
            a, na_t + a);
    }
    emitf(o, "  { double x = wsum(rcnt);\n"
             "    if ((tid & 63) == 0 && x != 0.0) atomicAdd(&bacc[%d], x); }\n",
          2 * na_t);
  }
  emitf(o, "  __syncthreads();\n"
           "  for (int i = tid; i < %d; i += WG)\n"
           "    out[(u64)blockIdx.x * %d + i] = bacc[i];\n"
           "}\n", nv, nv);
  return o;
}

/* compile (or fetch) the kernel for this plan; returns NULL on any failure
 * (caller falls back to the interpreted kernels) */
extern "C" void *sn_jit_get(void *cache, const sn_dev_plan *p,
                            const int *kinds, int nslots, int na_t,
                            int has_del) {
  auto *jc = (JitCache *)cache;
  if (!jc) return nullptr;
  /* hash the plan SHAPE only — predicate bounds and aggregate
   * coefficients are tokenized (read from the device plan at run time),
   * so every literal value of the same shape reuses one compiled kernel
   * (the reference's tokenized plan cache) */
  sn_dev_plan shape = *p;
  /* pointers travel as kernel params; only their NULLness shapes the
   * source.  Capacity (hcap_log2) is tokenized, so grow-and-retry and
   * fresh workspaces reuse one compiled kernel. */
  shape.jkeys = shape.jkeys ? (const int64_t *)1 : nullptr;
  shape.jpayload = shape.jpayload ? (const int32_t *)1 : nullptr;
  shape.jlut = shape.jlut ? (const int32_t *)1 : nullptr;
  shape.hkeys = nullptr;
  shape.hacc = nullptr;
  shape.hflags = nullptr;
  shape.hcap_log2 = 0;
  /* radix stays in the hash as a BOOLEAN (it changes the source; the
   * partition shift itself is read from the plan at run time) and the
   * record-buffer geometry is tokenized */
  shape.precs = nullptr;
  shape.pcount = nullptr;
  shape.percap = 0;
  shape.radix = shape.radix ? 1 : 0;
  for (int i = 0; i < 2; i++) {
    shape.inp[i].bm = shape.inp[i].bm ? (const uint64_t *)1 : nullptr;
    shape.inp[i].list = shape.inp[i].list ? (const int64_t *)1 : nullptr;
    shape.inp[i].base = 0;
    shape.inp[i].nwords = 0;
    shape.inp[i].n = 0;
  }
  auto trivm = [](double a, double m) { return (a == 0.0 && m == 1.0) ? 1.0 : 0.0; };
  for (int i = 0; i < 8; i++) { shape.preds_d[i].lo = shape.preds_d[i].hi = 0.0; }
  for (int i = 0; i < 4; i++) { shape.preds_i[i].lo = shape.preds_i[i].hi = 0; }
  for (int a = 0; a < 12; a++) {
    /* keep only the per-factor triviality bit (it changes the source) */
    double t0 = trivm(shape.aggs[a].a0, shape.aggs[a].m0);
    double t1 = trivm(shape.aggs[a].a1, shape.aggs[a].m1);
    double t2 = trivm(shape.aggs[a].a2, shape.aggs[a].m2);
    shape.aggs[a].a0 = t0; shape.aggs[a].m0 = 0.0;
    shape.aggs[a].a1 = t1; shape.aggs[a].m1 = 0.0;
    shape.aggs[a].a2 = t2; shape.aggs[a].m2 = 0.0;
  }
  uint64_t h = 1469598103934665603ull;
  auto mix = [&](const void *d, size_t n) {
    const uint8_t *b = (const uint8_t *)d;
    for (size_t i = 0; i < n; i++) { h ^= b[i]; h *= 1099511628211ull; }
  };
  mix(&shape, sizeof(shape));
  mix(kinds, sizeof(int) * SN_DEV_MAX_COLS);
  mix(&nslots, 4); mix(&na_t, 4); mix(&has_del, 4);
  if (getenv("SN_DBG_JIT")) {
    fprintf(stderr, "[jit] h=%016llx na=%d ns=%d pac=%d sparse=%d ops=",
            (unsigned long long)h, p->naggs, nslots, p->pac, p->sparse);
    for (int a = 0; a < p->naggs; a++) fprintf(stderr, "%d", p->aggs[a].op);
    fprintf(stderr, " nf=");
    for (int a = 0; a < p->naggs; a++) fprintf(stderr, "%d", p->aggs[a].nf);
    fprintf(stderr, "\n");
  }
  {
    std::lock_guard<std::mutex> g(jc->mu);
    auto it = jc->fns.find(h);
    if (it != jc->fns.end()) return (void *)it->second.fn;
  }
  std::string src = gen_source(p, kinds, nslots, na_t, has_del);
  if (const char *dump = getenv("SN_JIT_DUMP")) {
    if (FILE *f = fopen(dump, "a")) {
      fputs(src.c_str(), f);
      fputs("\n/* ---- */\n", f);
      fclose(f);
    }
  }
  std::lock_guard<std::mutex> g(jc->mu);
  auto it = jc->fns.find(h);
  if (it != jc->fns.end()) return (void *)it->second.fn;

  hiprtcProgram prog;
  if (hiprtcCreateProgram(&prog, src.c_str(), "sn_jit.cu", 0, nullptr,
                          nullptr) != HIPRTC_SUCCESS)
    return nullptr;
  const char *opts[] = { "--offload-arch=gfx950", "-O3" };
  hiprtcResult crc = hiprtcCompileProgram(prog, 2, opts);
  if (crc != HIPRTC_SUCCESS) {
    size_t lsz = 0;
    hiprtcGetProgramLogSize(prog, &lsz);
    std::vector<char> log(lsz + 1, 0);
    if (lsz) hiprtcGetProgramLog(prog, log.data());
    fprintf(stderr, "[sn_jit] compile failed:\n%s\n", log.data());
    hiprtcDestroyProgram(&prog);
    jc->fns[h] = JitFn();   /* negative-cache */
    return nullptr;
  }
  size_t csz = 0;
  hiprtcGetCodeSize(prog, &csz);
  std::vector<char> code(csz);
  hiprtcGetCode(prog, code.data());
  hiprtcDestroyProgram(&prog);

  JitFn jf;
  if (hipModuleLoadData(&jf.mod, code.data()) != hipSuccess) return nullptr;
  if (hipModuleGetFunction(&jf.fn, jf.mod, "jit_scan") != hipSuccess) {
    (void)hipModuleUnload(jf.mod);
    return nullptr;
  }
  jc->fns[h] = jf;
  return (void *)jf.fn;
}

extern "C" int sn_jit_cache_count(void *cache) {
  auto *jc = (JitCache *)cache;
  if (!jc) return 0;
  std::lock_guard<std::mutex> g(jc->mu);
  return (int)jc->fns.size();
}

extern "C" int sn_jit_launch(void *fn, int grid,
                             const sn_dev_batch *batches,
                             const sn_dev_tile *tiles, int ntiles,
                             double *scratch, const int64_t *jkeys,
                             const int32_t *jpayload, const int32_t *jlut,
                             const sn_dev_plan *plan_dev, void *stream) {
  void *args[] = { (void *)&batches, (void *)&tiles, (void *)&ntiles,
                   (void *)&scratch, (void *)&jkeys, (void *)&jpayload,
                   (void *)&jlut, (void *)&plan_dev };
  hipError_t e = hipModuleLaunchKernel((hipFunction_t)fn, grid, 1, 1,
                                       256, 1, 1, 0, (hipStream_t)stream,
                                       args, nullptr);
  return (int)e;
}
