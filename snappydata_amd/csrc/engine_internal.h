/*
 * engine_internal.h — device-visible descriptors shared between the host
 * runtime (engine.cpp) and the HIP kernels (kernels.hip).
 *
 * The engine stores column batches in HBM in the reference's encoded byte
 * format and decodes INSIDE the scan kernel.  The host
 * pre-parses each blob's header once at sn_batch_put and keeps these
 * descriptors so the kernel never re-parses headers:
 *   - body pointer (fixed-width array / dictionary index array)
 *   - null bitset words + a per-64-row prefix-sum of null counts, so
 *     nonNullPosition = prefix[row/64] + popcount(word & mask) is O(1) per
 *     row on the GPU (the reference decodes sequentially on the JVM —
 *     NullableDecoder, ColumnEncoding.scala:1069-1135 — which has no
 *     data-parallel analogue, hence the auxiliary prefix array)
 *   - delete mask re-expressed as a row bitmap
 *   - update deltas (ColumnDeltaDecoder/UpdatedColumnDecoder semantics)
 *     merged on host into a per-column sorted patch list applied by the
 *     kernel during the scan (deltas are small by construction:
 *     ColumnMaxDeltaRows=10000, Literals.scala:138-146)
 */
#ifndef SN_ENGINE_INTERNAL_H
#define SN_ENGINE_INTERNAL_H

#include <stdint.h>
#include <stdlib.h>

#define SN_DEV_MAX_COLS 8      /* referenced columns per plan */

/* column value kinds as seen by the kernel */
enum {
  SN_K_F64 = 0, SN_K_I32 = 1, SN_K_I64 = 2, SN_K_F32 = 3,
  SN_K_DICT16 = 4,   /* int16 dictionary index (group cols) */
  SN_K_DICT32 = 5,   /* int32 dictionary index (BigDictionary) */
  SN_K_I16 = 6, SN_K_BOOLBIT = 7,
  SN_K_U8 = 8,       /* uncompressed boolean byte body */
  SN_K_S8 = 9,       /* uncompressed int8 */
  SN_K_RLE = 10,     /* run-length: host-built run-ends + values aux */
  SN_K_RLE_I64 = 11  /* run-length over an INT64 column: rle_vals carry the
                        RAW int64 bits memcpy'd into doubles (the LDS-image
                        convention for i64), not numeric doubles.
                        (Int-typed DICTIONARY columns have no kernel kind:
                        the host materializes their values into a plain
                        fixed-width body at put — decompress-on-put, like
                        the LZ4 wrapper.) */
};

typedef struct {
  const void     *body;        /* fixed-width values / dict index array */
  const uint64_t *nullw;       /* null bitset words (NULL if none) */
  const uint32_t *nullpfx;     /* nulls in words [0, w) — host-built */
  const int32_t  *dictmap;     /* group col: local dict idx -> premultiplied
                                  global group contribution */
  /* update-delta patches (host-merged, sorted by row) */
  const uint64_t *patch_bm;    /* bitmap: row has patch */
  const int32_t  *patch_pos;   /* sorted patched rows */
  const double   *patch_val;   /* patched value (f64 or int bits in .i64) */
  const uint64_t *patch_nullbm;/* bitmap over patch index: patch writes NULL */
  /* RLE aux (host-extracted): cumulative run END positions (exclusive) and
   * run values widened to f64 (i64 raw-bitcast); kernel binary-searches the
   * run for each nonNullPosition (RunLengthEncoding.scala decodes
   * sequentially — no parallel analogue) */
  const int32_t *rle_ends;
  const double  *rle_vals;
  int32_t rle_n;
  int32_t patch_n;
  int32_t kind;
  int32_t has_nulls;
  int32_t null_gid;            /* group col: premultiplied global id of NULL */
} sn_dev_col;

typedef struct {
  int32_t num_rows;
  int32_t clean;               /* 1: no nulls/deletes/patches on any referenced
                                  col and all plain fixed-width -> fast path */
  const uint64_t *del_bm;      /* delete bitmap (NULL if none) */
  sn_dev_col cols[SN_DEV_MAX_COLS];
} sn_dev_batch;

typedef struct { int32_t batch; int32_t row_start; } sn_dev_tile;

/* Canonical (branchless) predicate: alive &= (lo <= x && x <= hi).
 * Strictness and missing bounds are folded by the host (nextafter for
 * strict double bounds, +-1 for strict integer bounds, +-inf when absent),
 * so the kernel does exactly two compares per predicate. */
typedef struct { double lo, hi; int32_t cslot; int32_t _p; } sn_dev_pred_d;
typedef struct { int64_t lo, hi; int32_t cslot; int32_t _p; } sn_dev_pred_i;

/* Canonical aggregate input: value = (a0+m0*x0)*(a1+m1*x1)*(a2+m2*x2) with
 * unused factors neutralized to (1 + 0*x_c0) — three fmas, no branches.
 * COUNT(*) is all-neutral (value 1).  nf/c-slots retained for the general
 * path's per-factor null checks.  op: 0 = sum-accumulate (SUM/AVG/COUNT),
 * 1 = MIN, 2 = MAX — min/max accumulators store the ORDER-PRESERVING u64
 * encoding (sn_f64_ord) so device atomics are integer atomicMin/Max;
 * k_reduce (dense) or the host (sparse readback) decodes. */
typedef struct {
  double a0, m0, a1, m1, a2, m2;
  int32_t c0, c1, c2;
  int32_t nf;                  /* real factors (0 for COUNT(*)) */
  int32_t op;
  int32_t _p2;
} sn_dev_agg;

/* order-preserving bijection double -> u64 (non-NaN): u64 min/max of the
 * encoding equals double min/max.  Identity elements: min = enc(+inf),
 * max = enc(-inf). */
static inline unsigned long long sn_f64_ord_h(double x) {
  unsigned long long b;
  __builtin_memcpy(&b, &x, 8);
  return (b & 0x8000000000000000ull) ? ~b : (b | 0x8000000000000000ull);
}
static inline double sn_ord_f64_h(unsigned long long o) {
  unsigned long long b = (o & 0x8000000000000000ull)
      ? (o ^ 0x8000000000000000ull) : ~o;
  double d;
  __builtin_memcpy(&d, &b, 8);
  return d;
}

typedef struct {
  int32_t npreds_d, npreds_i, naggs, ngroup;
  int32_t nslots, nused;
  uint32_t i64_mask;           /* bit c: cslot c is INT64 (raw bits in LDS) */
  int32_t gcol[2];
  /* broadcast-dimension probe (empty-slot sentinel key = INT64_MIN) */
  const int64_t *jkeys;        /* open-address key array (NULL = no join) */
  const int32_t *jpayload;     /* per-slot payload: dim attr gid (or 0) */
  int32_t jcap_log2;           /* table capacity = 1 << jcap_log2 */
  int32_t jcslot;              /* fact key column slot */
  int32_t jmode;               /* 0 semi, 1 group-by-dim-attr */
  int32_t jslot_mul;           /* >0: composite GROUP BY dim_attr, fact_col —
                                  slot = pay * jslot_mul + dense fact slot */
  /* dense probe LUT (built when the dim key span fits 16M entries):
   * pay = key in [jlut_min, jlut_max] ? jlut[key - jlut_min] : -1.
   * One dependent load per row instead of the open-address chain. */
  const int32_t *jlut;
  int64_t jlut_min, jlut_max;
  /* group slot formula: slot = ((v0 - gbase[0]) * gmul0) + (v1 - gbase[1]).
   * Dictionary key columns arrive premultiplied via dictmap (base 0, and
   * gmul0 folded into the map), integer key columns use their stats-derived
   * minimum as base and the second key's span as gmul0. */
  int64_t gbase[2];
  int32_t gmul0, _pad2;
  sn_dev_pred_d preds_d[8];
  sn_dev_pred_i preds_i[4];
  sn_dev_agg aggs[12];
  /* IN-list membership predicates: bitmap LUT over [base, base+nwords*64)
   * when the value span is dense enough (dictionary ids always are), else
   * a sorted list binary-searched per row (interpreted kernels only) */
  struct {
    const uint64_t *bm;
    const int64_t *list;
    int64_t base;
    int32_t nwords;
    int32_t n;
    int32_t cslot;
    int32_t _p;
  } inp[2];
  int32_t npreds_in;
  int32_t _pad4b;
  /* sparse-key open-address hash aggregate (the ByteBufferHashMap /
   * SHAMapAccessor analogue, ByteBufferHashMap.scala:140-183,
   * SHAMapAccessor.scala:716-830): integer group keys WITHOUT dense-slot
   * structure (int64 keys; int32/int16 spans beyond the dense cap or
   * without stats).  hkeys = open-address key array of 1<<hcap_log2
   * (sentinel SN_HASH_EMPTY = -1; a row whose key IS -1 lands in the
   * reserved row at index 1<<hcap_log2); hacc = accumulators
   * [(1<<hcap_log2)+1][naggs+1]; hflags[0] = overflow (table full — host
   * grows and relaunches).  ngroup==2 packs two 32-bit keys into one i64. */
  long long *hkeys;
  double *hacc;
  int32_t *hflags;
  int32_t hcap_log2;
  int32_t sparse;              /* 1: hash-aggregate mode */
  /* 1: per-agg counts — grouped accumulator rows widen to
   * [sums naggs][counts naggs][rowcount] because some aggregate input
   * column carries ACTUAL nulls (Spark Sum/Average skip null inputs,
   * SnappyHashAggregateExec.scala:450-500); 0 keeps [sums][rowcount] with
   * counts == rowcount (non-null inputs) */
  int32_t pac;
  int32_t _pad3;
  /* radix-partitioned two-pass hash aggregate (query-compiled pass 1 only;
   * big tables, clean batches).  Pass 1 scatters (key, agg values) records
   * into 1 << (hcap_log2 - SN_RADIX_SUB_LOG2) hash partitions instead of
   * probing the table; pass 2 (k_radix_agg) aggregates each partition into
   * its own 4096-slot SEGMENT of hkeys/hacc, so the probe + accumulate
   * working set per workgroup is ~100 KB and stays in the XCD's L2 instead
   * of thrashing HBM with 64 B random lines.  precs = record buffer
   * [npart][percap][(1+naggs) doubles] (slot 0 = key bits); pcount[npart]
   * fill counters; hflags[3] = partition overflow (host retries without
   * radix).  radix == 0 leaves the single-pass probe behavior. */
  double *precs;
  int32_t *pcount;
  int32_t percap;
  int32_t radix;
} sn_dev_plan;

/* radix partition shift (plan->radix): slots per partition table.
 * Swept on the 1M-key bench: sub 9 (512-slot LDS tables, 13 KB, high
 * occupancy) beat 10/11/12/13 at 25.7 / 24.9 / 21.2 / 17.0 / 6.5 Grows/s
 * — pass 2 wants occupancy, not bigger tables.  Clamped below so
 * npart = cap >> radix never exceeds the 4096 the pass-1 histogram
 * sizes for. */
#define SN_RADIX_SUB_MIN 9
#define SN_RADIX_NPART_MAX_LOG2 12
#define SN_RADIX_LDS_MAX (144u * 1024)
static inline unsigned sn_radix_lds_bytes(int sub_log2, int naggs1) {
  return (unsigned)((1u << sub_log2) * (8u + 8u * (unsigned)naggs1) +
                    256 /*WG*/ * 4u);
}
/* does pass 2 compact DIRECTLY from LDS (host then skips k_hash_compact)?
 * SN_RADIX_GLOB=1 forces the wide-row global-segment variant so its parity
 * is testable at small sizes (it otherwise only triggers at cap 2^24 with
 * 4 aggregates).  Shared by the host (compact decision) and the launcher
 * (kernel choice) so the two can never disagree. */
static inline int sn_radix_direct(int sub_log2, int naggs1) {
  const char *fg = getenv("SN_RADIX_GLOB");
  if (fg && fg[0] == '1') return 0;
  return sn_radix_lds_bytes(sub_log2, naggs1) <= SN_RADIX_LDS_MAX;
}

/* sparse hash-aggregate empty-slot sentinel: -1 so the host can memset the
 * key array (a REAL key of -1 routes to the reserved row at index cap) */
#define SN_HASH_EMPTY (-1ll)

#define SN_SCAN_CHUNK 1024     /* rows per LDS conversion chunk (kernels.hip) */

/* shared host/launcher routing predicate: does the grouped LDS-accumulator
 * kernel fit, or must the plan take the global-atomic route?  Used by BOTH
 * sn_launch_scan_agg and the engine (which must zero the global accumulator
 * before launch) so the two can never disagree. */
static inline int sn_grouped_needs_global(int nused, int nslots, int na1) {
  unsigned long long lds =
      (unsigned long long)nused * SN_SCAN_CHUNK * 8 +
      (unsigned long long)nused * (SN_SCAN_CHUNK / 64) * 8 +
      2ull * (SN_SCAN_CHUNK / 64) * 8 + sizeof(sn_dev_plan) + 512 +
      (SN_SCAN_CHUNK / 64) * 8 + SN_SCAN_CHUNK * 2 +
      (unsigned long long)nslots * na1 * 8 + 64;
  return lds > 160ull * 1024;
}

#define SN_GRID_CAP 2048
#define SN_RESULT_PAGE 1024      /* == SN_MAX_GROUP_SLOTS (result page) */
#define SN_BIG_GROUP_CAP (1 << 20) /* dense-slot cap of the global-atomic
                                      grouped path (>SN_RESULT_PAGE) */
#define SN_GRID_BIGSLOT 256   /* grid cap for the >16-slot LDS hash-agg path */
#define SN_TILE_ROWS 16384     /* rows per workgroup tile (16 LDS chunks;
                                  long pipelines for the staged conversion) */

#ifdef __cplusplus
extern "C" {
#endif

/* launches the scan+filter+aggregate over all tiles.
 * out: device array [nslots][naggs+1] doubles (last = group row count),
 * zeroed by caller.  Returns hipError_t as int. */
int sn_launch_scan_agg(const sn_dev_plan *plan,
                       const sn_dev_plan *dev_plan,  /* device copy (LDS mirror source) */
                       const sn_dev_batch *dev_batches,
                       const sn_dev_tile *dev_tiles, int32_t ntiles,
                       double *dev_out,
                       double *dev_scratch,  /* >= min(ntiles,SN_GRID_CAP) x nv */
                       void *stream);

/* put-time patch materialization into a null-free fixed-width device body */
int sn_launch_patch_apply(void *body, const int32_t *pos, const double *val,
                          int n, int kind, void *stream);

/* sparse-key open-address hash aggregate: scan into plan->hkeys/hacc, then
 * compact the used rows into (key, accumulator-row) pairs for readback */
int sn_launch_hash_scan(const sn_dev_plan *plan, const sn_dev_plan *dev_plan,
                        const sn_dev_batch *dev_batches,
                        const sn_dev_tile *dev_tiles, int32_t ntiles,
                        void *stream);
int sn_launch_hash_compact(const long long *hk, const double *hacc,
                           int cap, int naggs1, long long *okeys,
                           double *orows, int *counter, void *stream);

/* radix pass 2: aggregate each partition's records in an LDS table and
 * compact straight into (okeys, orows) — or, for rows too wide for LDS,
 * into per-partition segments of the global table (host runs
 * k_hash_compact as usual; sn_radix_direct says which) */
int sn_launch_radix_agg(const sn_dev_plan *plan, const sn_dev_plan *dev_plan,
                        long long *okeys, double *orows, int *counter,
                        void *stream);

/* device-side join-table build from a column table (colocated
 * partitioned-partitioned join): fills an open-address (key, payload)
 * table in the probe_sweep layout; optional LUT densification */
int sn_launch_join_build(const sn_dev_plan *plan, const sn_dev_plan *dev_plan,
                         const sn_dev_batch *dev_batches,
                         const sn_dev_tile *dev_tiles, int32_t ntiles,
                         long long *hk, int32_t *hp, int cap_log2,
                         int has_attr, int32_t *flags, void *stream);
int sn_launch_hash_to_lut(const long long *hk, const int32_t *hp,
                          long long cap, int32_t *lut, long long lmin,
                          void *stream);

/* wave-cooperative raw LZ4 block decode (f1 compressed-upload ingest);
 * err[0]: 0 ok, 1 malformed, 2 length mismatch */
int sn_launch_lz4_decompress(const void *src, long long slen, void *dst,
                             long long dlen, int32_t *err, void *stream);

/* f2: device min/max over one raw fixed-width column; omin memset 0xFF,
 * omax memset 0x00 (values land as order-preserving u64 encodings:
 * f64 ord / integer sign-bias) */
int sn_launch_col_minmax(const void *body, long long n, int kind,
                         unsigned long long *omin, unsigned long long *omax,
                         void *stream);

/* launches only the partial-fold (k_reduce): scratch[nblocks][nv] -> out.
 * Used by the JIT path, whose scan kernel writes the same scratch rows.
 * plan_dev (nullable) supplies per-agg ops for MIN/MAX folding. */
int sn_launch_reduce(const double *dev_scratch, int nblocks, int nv,
                     double *dev_out, int naggs1, int out_stride,
                     const void *plan_dev, void *stream);

/* initialize a global accumulator's MIN/MAX cells to their ord-encoding
 * identities (rows x na1 cells; sum/count cells keep the host's memset 0) */
int sn_launch_acc_init(double *acc, long long rows, int naggs, int na1,
                       const void *plan_dev, void *stream);

/* query-compiled scan kernels (jit.cpp, hipRTC).  sn_jit_get returns an
 * opaque hipFunction_t for the plan (compiling + caching on first use) or
 * NULL -> caller falls back to the interpreted kernels. */
void *sn_jit_cache_create(void);
void sn_jit_cache_destroy(void *cache);
int sn_jit_cache_count(void *cache);
void *sn_jit_get(void *cache, const sn_dev_plan *p, const int *kinds,
                 int nslots, int na_t, int has_del);
int sn_jit_launch(void *fn, int grid, const sn_dev_batch *batches,
                  const sn_dev_tile *tiles, int ntiles, double *scratch,
                  const int64_t *jkeys, const int32_t *jpayload,
                  const int32_t *jlut, const sn_dev_plan *plan_dev,
                  void *stream);

#ifdef __cplusplus
}
#endif
#endif
