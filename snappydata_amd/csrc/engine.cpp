/*
 * engine.cpp — host runtime of the MI355X-native columnar scan/filter/agg
 * engine behind the C ABI declared in include/snappy_engine.h.
 *
 * Responsibilities (each mirrors a reference seam — SnappyDataInc/snappydata):
 *  - batch registry keyed by (uuid, bucket): replaces the GemFire column
 *    region entries (ColumnFormatKey/ColumnFormatValue,
 *    ColumnFormatEntry.scala:99-102) with an in-process table of
 *    device-resident batches; sn_batch_put = ExternalStore.storeColumnBatch
 *    (ExternalStore.scala:43-45) fused with upload-to-HBM.
 *  - blob header pre-parse + auxiliary decode indexes (null-count prefix per
 *    64-row word, delete bitmap, host-merged update patches) so the GPU scan
 *    decodes in O(1) per row — see engine_internal.h.
 *  - per-column global dictionary interning across batches: the reference
 *    consumes batch-local dictionary indexes inside one generated loop
 *    (DictionaryEncoding.scala:85-160, DictionaryOptimizedMapAccessor);
 *    a data-parallel engine needs batch-local -> global group mapping, built
 *    here and shipped to the kernel as a tiny per-batch map array.
 *  - host-side stats-predicate batch skip (ColumnTableScan.generateStatPredicate
 *    :820-963): conservative, never skips batches with deltas/deletes.
 *  - thin planner/validator for the hot-path plan shapes
 *    (SnappyStrategies.scala:340,547-603).
 *  - the product batch builder (ColumnBatchCreator.createAndStoreBatch,
 *    ColumnBatchCreator.scala:46-131): sn_ingest_columns encodes raw arrays
 *    into reference-format blobs (uncompressed numerics, dictionary strings —
 *    the default encoder choice of ColumnEncoding.getColumnEncoder :838-870)
 *    with a ColumnStatsSchema stats row, and a seeded TPC-H lineitem
 *    generator for benchmarks (no network: synthetic data).
 *
 * The GPU path NEVER falls back to CPU: any query on an engine without a HIP
 * device fails with SN_ERR_NOGPU.
 */
#include <hip/hip_runtime.h>
#include <dlfcn.h>

#include <algorithm>
#include <atomic>
#include <climits>
#include <cmath>
#include <cstdarg>
#include <cstdio>
#include <cstring>
#include <malloc.h>
#include <map>
#include <set>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "../../include/snappy_engine.h"
#include "engine_internal.h"

/* ------------------------------------------------------------------ */
/* error reporting                                                     */
static thread_local char g_err[512];
static int fail(int code, const char *fmt, ...) {
  va_list ap; va_start(ap, fmt);
  vsnprintf(g_err, sizeof(g_err), fmt, ap);
  va_end(ap);
  return code;
}
extern "C" const char *sn_last_error(void) { return g_err; }
extern "C" const char *sn_engine_arch(void) { return "gfx950"; }

#define HIP_OR_FAIL(x) do { hipError_t e_ = (x); if (e_ != hipSuccess) \
  return fail(SN_ERR_GENERIC, "%s failed: %s", #x, hipGetErrorString(e_)); } while (0)

/* ------------------------------------------------------------------ */
/* little-endian readers (host is LE)                                  */
static inline int32_t rd_i32(const uint8_t *p) { int32_t v; memcpy(&v, p, 4); return v; }
static inline int64_t rd_i64(const uint8_t *p) { int64_t v; memcpy(&v, p, 8); return v; }
static inline double  rd_f64(const uint8_t *p) { double v; memcpy(&v, p, 8); return v; }
static inline float   rd_f32(const uint8_t *p) { float v; memcpy(&v, p, 4); return v; }
static inline int16_t rd_i16(const uint8_t *p) { int16_t v; memcpy(&v, p, 2); return v; }

/* ------------------------------------------------------------------ */
/* device memory arena: bump allocator over big HBM slabs, plus a
 * size-bucketed free list so per-query transients (result buffers,
 * uncached descriptor sets, sparse plan copies) recycle instead of
 * leaking ~100 KB/query in a long-lived executor */
struct Arena {
  int device = -1;               /* -1: host-only shadow mode */
  std::vector<void *> slabs;
  std::vector<void *> host_allocs;   /* host-only mode: freed on destroy */
  size_t slab_sz = 512ull << 20;
  size_t off = 0;
  std::multimap<size_t, void *> freelist;   /* rounded size -> block */
  std::mutex mu;

  static size_t rnd(size_t n) { return (n + 255) & ~size_t(255); }

  void *alloc(size_t n) {
    std::lock_guard<std::mutex> g(mu);
    n = rnd(n);
    if (device < 0) {
      void *p = malloc(n);
      if (p) host_allocs.push_back(p);
      return p;
    }
    /* exact-bucket reuse (blocks are only ever freed at the size they
     * were allocated with, so exact match is the common case) */
    auto it = freelist.find(n);
    if (it != freelist.end()) {
      void *p = it->second;
      freelist.erase(it);
      return p;
    }
    if (slabs.empty() || off + n > slab_sz) {
      size_t sz = std::max(slab_sz, n);
      void *p = nullptr;
      if (hipMalloc(&p, sz) != hipSuccess) return nullptr;
      slabs.push_back(p);
      off = 0;
      if (sz != slab_sz) { off = n; return p; }
    }
    void *p = (char *)slabs.back() + off;
    off += n;
    return p;
  }
  /* return a block for reuse (device mode only; host blocks die with the
   * arena).  n must be the original request size. */
  void release(void *p, size_t n) {
    if (!p || device < 0) return;
    std::lock_guard<std::mutex> g(mu);
    freelist.emplace(rnd(n), p);
  }
  ~Arena() {
    for (void *p : slabs) (void)hipFree(p);
    for (void *p : host_allocs) free(p);
  }
};

/* ------------------------------------------------------------------ */
/* parsed column blob metadata (host view)                             */
struct ColMeta {
  int type_id = 0;               /* encoding */
  int64_t body_off = 0;          /* offset of fixed-width body / index array */
  int64_t null_off = 0;          /* offset of null words (0 if none) */
  int32_t num_null_words = 0;
  int32_t dict_n = 0;            /* dictionary entries (dict encodings) */
  std::vector<std::string> dict; /* string dictionary values */
  std::vector<int64_t> dict_i;   /* int dictionary values */
  std::vector<int32_t> local2global;  /* per-batch dict idx -> table-global id */
};

/* host-merged update patch for one column of one batch */
struct Patch {
  std::vector<int32_t> pos;      /* sorted row ordinals */
  std::vector<double> val;       /* f64 value or integer bits as double;
                                    dict cols: raw GLOBAL dict id */
  std::vector<uint8_t> isnull;
};

struct Batch {
  int64_t uuid = 0;
  int32_t bucket = 0;
  int32_t num_rows = 0;
  bool has_deltas = false;
  bool has_deletes = false;
  /* device pointers */
  std::vector<void *> col_dev;        /* whole blob per column */
  std::vector<const uint32_t *> nullpfx_dev;
  const uint64_t *del_bm_dev = nullptr;
  /* per-col patch device data */
  struct PatchDev {
    const uint64_t *bm = nullptr;
    const int32_t *pos = nullptr;
    const double *val = nullptr;
    const uint64_t *nullbm = nullptr;
    int32_t n = 0;
  };
  std::vector<PatchDev> patch_dev;
  std::vector<Patch> patch_host;      /* kept for dict premultiply at query */
  std::vector<uint8_t> had_patches;   /* survives materialization: int group
                                         keys must reject patched columns
                                         (stats no longer bound the values) */
  /* RLE aux (device): cumulative run ends + values widened to f64 */
  std::vector<const int32_t *> rle_ends_dev;
  std::vector<const double *> rle_vals_dev;
  std::vector<int32_t> rle_n;
  /* premultiplied dictionary maps, cached per (mul, null_gid): profiles
   * showed ~2000 per-batch map uploads on a first Q1 SF=100 submit —
   * local2global is fixed per batch, so the device map is reusable
   * across queries with the same group geometry */
  std::vector<std::map<std::pair<int32_t, int32_t>, const int32_t *>> dictmap_cache;
  /* f2 raw ingest: device-computed stats pending a sync.  Layout:
   * [nc mins][nc maxs] as order-preserving u64 encodings (f64 ord / int
   * sign-bias); is_raw marks which columns the device bounds cover. */
  const unsigned long long *stats_dev = nullptr;
  std::vector<uint8_t> is_raw;
  bool stats_pending = false;
  std::vector<ColMeta> cols;
  /* stats (parsed) */
  bool stats_valid = false;
  std::vector<double> lo_d, hi_d;
  std::vector<int64_t> lo_i, hi_i;
  std::vector<int32_t> null_count;
  std::vector<uint8_t> bounds_null;
  /* host shadow (device == -1 only) */
  std::vector<std::vector<uint8_t>> host_blobs;
};

/* cached device descriptor set for one plan shape (the reference caches
 * generated plans per query signature, SnappySession plan cache) */
struct DescCache {
  uint64_t sig = 0;
  size_t batch_count = 0;       /* invalidation: table grew */
  const void *db_dev = nullptr;
  const void *tl_dev = nullptr;
  size_t db_bytes = 0, tl_bytes = 0;   /* released to the free list on evict */
  int32_t ntiles = 0;
  int64_t rows = 0;
  /* JIT eligibility of this descriptor set: every batch clean with these
   * uniform stageable kinds (needed again on cache hits, when the
   * per-batch loop that derives them is skipped) */
  int jit_ok = 0;
  int jit_del = 0;
  int jit_kinds[SN_DEV_MAX_COLS] = {0};
  int pac = 0;        /* batches carry nulls on aggregate-input columns */
};

struct Table {
  std::string name;
  std::vector<sn_col_schema> schema;
  std::vector<Batch> batches;
  std::vector<DescCache> desc_caches;
  /* global dictionary per column (string dict cols) */
  std::vector<std::vector<std::string>> gdict;
  std::vector<std::map<std::string, int32_t>> gdict_idx;
  /* longest interned entry per column: group keys beyond SN_KEY_MAX-1
   * bytes cannot travel in results/partial blocks — queries grouping on
   * such a column fail loudly instead of silently merging truncated keys */
  std::vector<int32_t> gdict_maxlen;
  /* last adequate sparse hash-table capacity (log2): later queries start
   * there instead of rediscovering it through grow-and-retry */
  int sparse_cap_hint = 0;
  int64_t total_rows = 0;
  std::mutex mu;
};

/* broadcast dimension (row-store stand-in): host rows + cached device
 * open-address table (HashedObjectCache analogue, HashJoinExec.scala:449-470) */
struct Dim {
  std::string name;
  std::vector<int64_t> keys;
  std::vector<std::string> attrs;          /* per-key attr (may be empty) */
  std::vector<std::string> attr_dict;      /* distinct attrs (gid order) */
  std::vector<int32_t> attr_gid;           /* per-key gid */
  /* device table (built lazily, cached) */
  const int64_t *dev_keys = nullptr;
  const int32_t *dev_payload = nullptr;
  int32_t cap_log2 = 0;
  /* dense payload LUT over [lut_min, lut_max] when the span is small */
  const int32_t *dev_lut = nullptr;
  int64_t lut_min = 0, lut_max = -1;
  int32_t attr_maxlen = 0;
  bool from_table = false;   /* device-built from a column table (f3):
                                host keys unknown, sn_dim_put forbidden */
  /* guards keys/attrs/device-table pointers: sn_dim_put may run while a
   * submit builds/reads the device table (the reference's replicated
   * region puts are similarly concurrent with task-side get()s) */
  std::mutex mu;
};

struct sn_engine {
  sn_config cfg;
  Arena arena;
  std::vector<std::unique_ptr<Table>> tables;
  std::vector<std::unique_ptr<Dim>> dims;
  hipStream_t stream = nullptr;
  bool has_gpu = false;
  /* reusable per-engine scratch for block-partial reduction rows
   * (queries on one engine serialize on its stream) */
  double *scratch = nullptr;
  size_t scratch_sz = 0;
  /* sparse hash-aggregate workspace (reused: the sparse path reads its
   * results back inside submit, so no two queries share it live) */
  long long *hws_keys = nullptr;
  long long *hws_okeys = nullptr;
  double *hws_acc = nullptr;
  double *hws_orows = nullptr;
  int32_t *hws_flags = nullptr;   /* [0] overflow, [1] compact counter,
                                     [2] fill, [3] radix record overflow */
  int hws_cap_log2 = 0;
  size_t hws_acc_bytes = 0;
  /* radix two-pass record buffer + per-partition fill counters (engine-
   * cached like the table workspace; sized to the largest query seen) */
  double *rws_recs = nullptr;
  size_t rws_bytes = 0;
  int32_t *rws_pcount = nullptr;
  int rws_npart = 0;
  int32_t *lz4_err_dev = nullptr; /* device error word for the LZ4 decode */
  std::mutex lz4_mu;              /* serializes decode launch+sync+readback */
  /* serializes sn_query_submit: queries on one engine share the stream,
   * the block-partial scratch and the hash/radix workspaces, so two
   * submits from different threads (even on DIFFERENT tables — t->mu
   * does not cover this) must not interleave their enqueue+readback
   * sections.  Ingest does not take this lock. */
  std::mutex query_mu;
  /* pinned staging for big blob uploads: pageable hipMemcpy bounces through
   * the runtime's staging path at ~1-2 GB/s with a device sync per call
   * (measured: SF=10 ingest spent ~2 s there, 27 Mrows/s).  A pool of
   * pinned bounce slots lets concurrent ingest threads overlap their
   * host->pinned memcpys while the DMAs queue at full PCIe rate
   * (one slot: 105 Mrows/s measured; the pool removes the memcpy serial). */
  static const int PIN_SLOTS = 4;
  struct PinSlot { void *buf = nullptr; size_t sz = 0; std::mutex mu; };
  PinSlot pins[PIN_SLOTS];
  std::atomic<uint32_t> pin_rr { 0 };
  /* per-engine cache of query-compiled kernels (jit.cpp) */
  void *jit = nullptr;
  /* steady-state submit caches: device plan copies by content hash, and a
   * small HIP-event pool (create/destroy costs ~µs per query otherwise) */
  std::map<uint64_t, void *> dp_cache;
  std::vector<hipEvent_t> ev_pool;
  std::set<sn_query *> live_q;   /* outstanding queries: detached (e=NULL)
                                    at engine destroy so a later
                                    sn_query_destroy stays safe */
  std::mutex aux_mu;
  std::mutex mu;

  hipEvent_t ev_acquire() {
    { std::lock_guard<std::mutex> g(aux_mu);
      if (!ev_pool.empty()) { hipEvent_t e2 = ev_pool.back(); ev_pool.pop_back(); return e2; } }
    hipEvent_t ev = nullptr;
    (void)hipEventCreate(&ev);
    return ev;
  }
  void ev_release(hipEvent_t ev) {
    if (!ev) return;
    std::lock_guard<std::mutex> g(aux_mu);
    if (ev_pool.size() < 16) ev_pool.push_back(ev);
    else (void)hipEventDestroy(ev);
  }
};

/* ------------------------------------------------------------------ */
extern "C" sn_engine *sn_engine_create(const sn_config *cfg) {
  auto *e = new sn_engine();
  if (cfg) e->cfg = *cfg;
  else memset(&e->cfg, 0, sizeof(e->cfg));
  if (e->cfg.column_batch_size <= 0) e->cfg.column_batch_size = 24ll << 20;
  if (e->cfg.column_max_delta_rows <= 0) e->cfg.column_max_delta_rows = 10000;
  if (e->cfg.hash_join_size <= 0) e->cfg.hash_join_size = 100ll << 20;
  if (e->cfg.shard_count <= 0) e->cfg.shard_count = 1;
  if (e->cfg.n_buckets <= 0) e->cfg.n_buckets = 128;
  int ndev = 0;
  if (hipGetDeviceCount(&ndev) == hipSuccess && ndev > 0 && e->cfg.device >= 0) {
    if (hipSetDevice(e->cfg.device) == hipSuccess &&
        hipStreamCreate(&e->stream) == hipSuccess) {
      e->has_gpu = true;
      e->arena.device = e->cfg.device;
    }
  }
  if (!e->has_gpu) {
    e->arena.device = -1;   /* host-only: ingest/encode testable, queries fail */
  } else {
    e->jit = sn_jit_cache_create();
  }
  /* big per-query result vectors (1M-group readbacks are ~24 MB) must
   * recycle through the heap, not mmap/munmap per query — the page-fault
   * churn measured as a bimodal 1.4-3.6 ms of host time per sparse query */
  (void)mallopt(M_MMAP_THRESHOLD, 256 << 20);
  (void)mallopt(M_TRIM_THRESHOLD, 256 << 20);
  return e;
}

/* D2H copy through a pinned bounce slot: pageable hipMemcpy D2H runs at
 * ~50 GB/s with run-to-run stalls (the sparse 1M-group readback measured
 * 1.4-3.6 ms of host time per query on identical kernels); the pinned DMA
 * + one host memcpy is both faster and stable */
static hipError_t d2h_copy(sn_engine *e, void *dst, const void *src, size_t n) {
  if (n < (1u << 20))
    return hipMemcpy(dst, src, n, hipMemcpyDeviceToHost);
  auto &slot = e->pins[e->pin_rr.fetch_add(1) % sn_engine::PIN_SLOTS];
  std::lock_guard<std::mutex> g(slot.mu);
  if (slot.sz < n) {
    size_t want = std::max<size_t>(n, 32u << 20);
    void *p = nullptr;
    if (hipHostMalloc(&p, want) != hipSuccess)
      return hipMemcpy(dst, src, n, hipMemcpyDeviceToHost);  /* fall back */
    if (slot.buf) (void)hipHostFree(slot.buf);
    slot.buf = p;
    slot.sz = want;
  }
  hipError_t rc = hipMemcpy(slot.buf, src, n, hipMemcpyDeviceToHost);
  if (rc != hipSuccess) return rc;
  memcpy(dst, slot.buf, n);
  return hipSuccess;
}

/* H2D copy through a pinned bounce slot (big transfers; small ones direct) */
static hipError_t h2d_copy(sn_engine *e, void *dst, const void *src, size_t n) {
  if (n < (1u << 20))
    return hipMemcpy(dst, src, n, hipMemcpyHostToDevice);
  auto &slot = e->pins[e->pin_rr.fetch_add(1) % sn_engine::PIN_SLOTS];
  std::lock_guard<std::mutex> g(slot.mu);
  if (slot.sz < n) {
    size_t want = std::max<size_t>(n, 32u << 20);
    void *p = nullptr;
    if (hipHostMalloc(&p, want) != hipSuccess)
      return hipMemcpy(dst, src, n, hipMemcpyHostToDevice);  /* fall back */
    if (slot.buf) (void)hipHostFree(slot.buf);
    slot.buf = p;
    slot.sz = want;
  }
  memcpy(slot.buf, src, n);
  return hipMemcpy(dst, slot.buf, n, hipMemcpyHostToDevice);
}

static void sn_detach_queries(sn_engine *e);   /* defined below sn_query */

extern "C" void sn_engine_destroy(sn_engine *e) {
  if (!e) return;
  sn_detach_queries(e);
  for (auto &ps : e->pins)
    if (ps.buf) (void)hipHostFree(ps.buf);
  for (hipEvent_t ev : e->ev_pool) (void)hipEventDestroy(ev);
  if (e->jit) sn_jit_cache_destroy(e->jit);
  if (e->stream) (void)hipStreamDestroy(e->stream);
  delete e;
}

extern "C" int32_t sn_table_define(sn_engine *e, const char *name,
                                   int32_t ncols, const sn_col_schema *schema) {
  if (!e || ncols <= 0 || ncols > 64 || !schema) { fail(SN_ERR_BADARG, "bad table args"); return SN_ERR_BADARG; }
  std::lock_guard<std::mutex> g(e->mu);
  auto t = std::make_unique<Table>();
  t->name = name ? name : "";
  t->schema.assign(schema, schema + ncols);
  t->gdict.resize(ncols);
  t->gdict_idx.resize(ncols);
  t->gdict_maxlen.assign(ncols, 0);
  e->tables.push_back(std::move(t));
  return (int32_t)e->tables.size() - 1;
}

extern "C" int32_t sn_dim_define(sn_engine *e, const char *name) {
  if (!e) return SN_ERR_BADARG;
  std::lock_guard<std::mutex> g(e->mu);
  auto d = std::make_unique<Dim>();
  d->name = name ? name : "";
  e->dims.push_back(std::move(d));
  return (int32_t)e->dims.size() - 1;
}

extern "C" int32_t sn_dim_put(sn_engine *e, int32_t dim, int64_t nkeys,
                              const int64_t *keys, const char *attr_payload,
                              const int32_t *attr_lens) {
  if (!e || dim < 0 || dim >= (int32_t)e->dims.size() || !keys || nkeys <= 0)
    return fail(SN_ERR_BADARG, "bad dim args");
  Dim *d = e->dims[dim].get();
  std::lock_guard<std::mutex> gd(d->mu);
  if (d->from_table)
    return fail(SN_ERR_BADARG, "dimension was built from a column table");
  int64_t off = 0;
  for (int64_t i = 0; i < nkeys; i++) {
    d->keys.push_back(keys[i]);
    if (attr_payload && attr_lens) {
      std::string a(attr_payload + off, (size_t)attr_lens[i]);
      off += attr_lens[i];
      int32_t gid = -1;
      for (size_t j = 0; j < d->attr_dict.size(); j++)
        if (d->attr_dict[j] == a) { gid = (int32_t)j; break; }
      if (gid < 0) { gid = (int32_t)d->attr_dict.size(); d->attr_dict.push_back(a); }
      if ((int32_t)a.size() > d->attr_maxlen) d->attr_maxlen = (int32_t)a.size();
      d->attrs.push_back(std::move(a));
      d->attr_gid.push_back(gid);
    } else {
      d->attrs.emplace_back();
      d->attr_gid.push_back(0);
    }
  }
  /* invalidate the cached device table (and the dense LUT with it) */
  d->dev_keys = nullptr; d->dev_payload = nullptr;
  d->dev_lut = nullptr; d->lut_min = 0; d->lut_max = -1;
  return SN_OK;
}

static void *up(sn_engine *e, const void *host, size_t n);
static Table *get_table(sn_engine *e, int32_t t);
static void sync_pending_stats(sn_engine *e, Table *t);

static inline uint64_t mix64h(uint64_t x) {
  x += 0x9E3779B97f4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

/* ---- partitioned-partitioned (colocated) join build: populate a Dim from
 * a resident column table, built ON DEVICE (k_join_build) — the
 * HashJoinExec per-task ObjectHashSet build + HashedObjectCache reuse
 * (HashJoinExec.scala:285-520, :449-470).  Colocated tables join
 * bucket-locally, so each shard builds from its own build-side rows. ---- */
extern "C" int32_t sn_dim_from_table(sn_engine *e, int32_t dim, int32_t table,
                                     int32_t key_col, int32_t attr_col) {
  if (!e || dim < 0 || dim >= (int32_t)e->dims.size())
    return fail(SN_ERR_BADARG, "bad dim handle");
  Table *t = get_table(e, table);
  if (!t) return fail(SN_ERR_BADARG, "bad table");
  if (!e->has_gpu)
    return fail(SN_ERR_NOGPU, "join build runs on device");
  Dim *d = e->dims[dim].get();
  std::lock_guard<std::mutex> gd(d->mu);
  if (d->dev_keys || !d->keys.empty())
    return fail(SN_ERR_BADARG, "dimension is not empty");
  std::lock_guard<std::mutex> g(t->mu);
  const int nc = (int)t->schema.size();
  if (key_col < 0 || key_col >= nc) return fail(SN_ERR_BADARG, "bad key col");
  sn_type_t kdt = t->schema[key_col].dtype;
  if (kdt != SN_TYPE_INT32 && kdt != SN_TYPE_INT64 && kdt != SN_TYPE_INT16)
    return fail(SN_ERR_UNSUPPORTED, "join build key must be int16/int32/int64");
  const bool has_attr = attr_col >= 0;
  if (has_attr) {
    if (attr_col >= nc || t->schema[attr_col].dtype != SN_TYPE_STRING)
      return fail(SN_ERR_UNSUPPORTED, "attr col must be a string column");
    if (t->schema[attr_col].nullable)
      return fail(SN_ERR_UNSUPPORTED,
                  "nullable attribute columns not supported in join build");
    if (t->gdict_maxlen[attr_col] > SN_KEY_MAX - 1)
      return fail(SN_ERR_UNSUPPORTED, "attr exceeds the group-key limit");
  }
  /* (key columns may be nullable: null keys never join and are masked at
   * build time via the validity bitmap) */
  if (t->total_rows > (1ll << 23))
    return fail(SN_ERR_UNSUPPORTED,
                "build side exceeds 2^23 rows (HashJoinSize-class bound)");
  sync_pending_stats(e, t);   /* the LUT span reads stats bounds */

  sn_dev_plan dp;
  memset(&dp, 0, sizeof(dp));
  dp.nused = has_attr ? 2 : 1;
  dp.i64_mask = kdt == SN_TYPE_INT64 ? 1u : 0u;
  dp.naggs = 1;

  std::vector<sn_dev_batch> hb;
  std::vector<sn_dev_tile> ht;
  const int used[2] = { key_col, attr_col };
  for (auto &b : t->batches) {
    sn_dev_batch db;
    memset(&db, 0, sizeof(db));
    db.num_rows = b.num_rows;
    db.del_bm = b.del_bm_dev;
    bool clean = !b.has_deletes;
    for (int ui = 0; ui < dp.nused; ui++) {
      int c = used[ui];
      const ColMeta &m = b.cols[c];
      sn_dev_col &dc = db.cols[ui];
      memset(&dc, 0, sizeof(dc));
      sn_type_t dt = t->schema[c].dtype;
      dc.body = (const uint8_t *)b.col_dev[c] + m.body_off;
      dc.has_nulls = m.num_null_words > 0;
      if (dc.has_nulls) {
        dc.nullw = (const uint64_t *)((const uint8_t *)b.col_dev[c] + m.null_off);
        dc.nullpfx = b.nullpfx_dev[c];
        clean = false;
      }
      switch (m.type_id) {
        case SN_ENC_UNCOMPRESSED:
          switch (dt) {
            case SN_TYPE_INT32: dc.kind = SN_K_I32; break;
            case SN_TYPE_INT64: dc.kind = SN_K_I64; break;
            case SN_TYPE_INT16: dc.kind = SN_K_I16; break;
            default:
              return fail(SN_ERR_UNSUPPORTED, "build col %d dtype", c);
          }
          break;
        case SN_ENC_RUNLENGTH:
          if (b.rle_n[c] <= 0 && b.num_rows > 0)
            return fail(SN_ERR_UNSUPPORTED, "RLE col %d without run aux", c);
          dc.kind = dt == SN_TYPE_INT64 ? SN_K_RLE_I64 : SN_K_RLE;
          dc.rle_ends = b.rle_ends_dev[c];
          dc.rle_vals = b.rle_vals_dev[c];
          dc.rle_n = b.rle_n[c];
          break;
        case SN_ENC_DICTIONARY:
        case SN_ENC_BIG_DICTIONARY: {
          if (dt != SN_TYPE_STRING)
            return fail(SN_ERR_UNSUPPORTED, "unexpected dict col %d", c);
          dc.kind = m.type_id == SN_ENC_DICTIONARY ? SN_K_DICT16 : SN_K_DICT32;
          /* payload gid = table-global dictionary id (mul 1, null slot 0) */
          auto &mcache = b.dictmap_cache[c];
          auto mit = mcache.find({ 1, 0 });
          if (mit != mcache.end()) {
            dc.dictmap = mit->second;
          } else {
            std::vector<int32_t> map(m.local2global.size() + 1);
            for (size_t i = 0; i < m.local2global.size(); i++)
              map[i] = m.local2global[i];
            map[m.local2global.size()] = 0;
            dc.dictmap = (const int32_t *)up(e, map.data(), map.size() * 4);
            if (!dc.dictmap) return fail(SN_ERR_NOMEM, "dictmap upload");
            mcache.emplace(std::make_pair(1, 0), dc.dictmap);
          }
          dc.null_gid = 0;
          break;
        }
        default:
          return fail(SN_ERR_UNSUPPORTED, "encoding %d on build path", m.type_id);
      }
      if (b.patch_dev[c].n > 0) {
        clean = false;
        const Batch::PatchDev &pd = b.patch_dev[c];
        dc.patch_bm = pd.bm;
        dc.patch_pos = pd.pos;
        dc.patch_nullbm = pd.nullbm;
        dc.patch_n = pd.n;
        dc.patch_val = pd.val;   /* string patches carry global ids (mul 1) */
      }
    }
    db.clean = clean ? 1 : 0;
    int32_t bi = (int32_t)hb.size();
    hb.push_back(db);
    for (int32_t r = 0; r < b.num_rows; r += SN_TILE_ROWS)
      ht.push_back({ bi, r });
  }

  int cap_log2 = 10;
  while ((1ll << cap_log2) < 2 * t->total_rows && cap_log2 < 25) cap_log2++;
  const size_t cap = 1ull << cap_log2;
  long long *hk = (long long *)e->arena.alloc(cap * 8);
  int32_t *hp = (int32_t *)e->arena.alloc(cap * 4);
  int32_t *flags = (int32_t *)e->arena.alloc(64);
  void *dp_dev = e->arena.alloc(sizeof(dp));
  void *db_dev = hb.empty() ? nullptr : e->arena.alloc(hb.size() * sizeof(sn_dev_batch));
  void *tl_dev = ht.empty() ? nullptr : e->arena.alloc(ht.size() * sizeof(sn_dev_tile));
  if (!hk || !hp || !flags || !dp_dev || (!hb.empty() && (!db_dev || !tl_dev)))
    return fail(SN_ERR_NOMEM, "join build alloc");
  if (hipMemsetAsync(flags, 0, 16, e->stream) != hipSuccess ||
      hipMemcpy(dp_dev, &dp, sizeof(dp), hipMemcpyHostToDevice) != hipSuccess)
    return fail(SN_ERR_GENERIC, "join build upload");
  if (!hb.empty() &&
      (hipMemcpy(db_dev, hb.data(), hb.size() * sizeof(sn_dev_batch),
                 hipMemcpyHostToDevice) != hipSuccess ||
       hipMemcpy(tl_dev, ht.data(), ht.size() * sizeof(sn_dev_tile),
                 hipMemcpyHostToDevice) != hipSuccess))
    return fail(SN_ERR_GENERIC, "join build upload");
  int rc = sn_launch_join_build(&dp, (const sn_dev_plan *)dp_dev,
                                (const sn_dev_batch *)db_dev,
                                (const sn_dev_tile *)tl_dev,
                                (int32_t)ht.size(), hk, hp, cap_log2,
                                has_attr ? 1 : 0, flags, e->stream);
  if (rc != 0)
    return fail(SN_ERR_GENERIC, "join build: %s",
                hipGetErrorString((hipError_t)rc));
  if (hipStreamSynchronize(e->stream) != hipSuccess)
    return fail(SN_ERR_GENERIC, "join build sync");
  int32_t fl[2] = { 0, 0 };
  (void)hipMemcpy(fl, flags, 8, hipMemcpyDeviceToHost);
  if (fl[0])
    return fail(SN_ERR_BADARG,
                "duplicate build keys with conflicting payloads (or a "
                "sentinel-valued key)");
  if (fl[1]) return fail(SN_ERR_OVERFLOW, "join build table full");

  d->dev_keys = (const int64_t *)hk;
  d->dev_payload = hp;
  d->cap_log2 = cap_log2;
  d->from_table = true;
  if (has_attr) {
    d->attr_dict = t->gdict[attr_col];
    d->attr_maxlen = t->gdict_maxlen[attr_col];
  } else {
    d->attr_dict.assign(1, std::string());
  }
  /* dense LUT when the key span (from stats bounds) is small */
  bool have_span = !t->batches.empty();
  int64_t mn = INT64_MAX, mx = INT64_MIN;
  for (auto &b : t->batches) {
    if (!b.stats_valid || b.bounds_null.size() <= (size_t)key_col ||
        b.bounds_null[key_col]) { have_span = false; break; }
    mn = std::min(mn, b.lo_i[key_col]);
    mx = std::max(mx, b.hi_i[key_col]);
  }
  if (have_span && mx >= mn && mx - mn + 1 <= (1ll << 24)) {
    int64_t span = mx - mn + 1;
    int32_t *lut = (int32_t *)e->arena.alloc((size_t)span * 4);
    if (lut &&
        hipMemsetAsync(lut, 0xff, (size_t)span * 4, e->stream) == hipSuccess &&
        sn_launch_hash_to_lut(hk, hp, (long long)cap, lut, mn,
                              e->stream) == 0 &&
        hipStreamSynchronize(e->stream) == hipSuccess) {
      d->dev_lut = lut;
      d->lut_min = mn;
      d->lut_max = mx;
    }
  }
  return SN_OK;
}

/* build (or reuse) the device open-address table: the broadcast into HBM */
static int dim_device_table(sn_engine *e, Dim *d) {
  std::lock_guard<std::mutex> gd(d->mu);
  if (d->dev_keys) return SN_OK;
  int32_t lg = 1;
  while ((1u << lg) < 2 * d->keys.size() + 1) lg++;
  size_t cap = 1ull << lg;
  std::vector<int64_t> hk(cap, INT64_MIN);
  std::vector<int32_t> hp(cap, -1);
  for (size_t i = 0; i < d->keys.size(); i++) {
    int64_t k = d->keys[i];
    uint32_t h = (uint32_t)mix64h((uint64_t)k) & (cap - 1);
    while (hk[h] != INT64_MIN) {
      if (hk[h] == k) return fail(SN_ERR_BADARG, "duplicate dimension key");
      h = (h + 1) & (cap - 1);
    }
    hk[h] = k;
    hp[h] = d->attr_gid[i];
  }
  d->dev_keys = (const int64_t *)up(e, hk.data(), cap * 8);
  d->dev_payload = (const int32_t *)up(e, hp.data(), cap * 4);
  d->cap_log2 = lg;
  if (!d->dev_keys || !d->dev_payload) return fail(SN_ERR_NOMEM, "dim upload");
  /* dense LUT when the key span is small (16M entries = 64 MB worst case;
   * typical dimensions are far smaller and sit in L2): replaces the
   * dependent open-address chain with one load per probed row */
  if (!d->keys.empty()) {
    int64_t mn = d->keys[0], mx = d->keys[0];
    for (int64_t k : d->keys) { mn = k < mn ? k : mn; mx = k > mx ? k : mx; }
    int64_t span = mx - mn + 1;
    if (span > 0 && span <= (1ll << 24)) {
      std::vector<int32_t> lut((size_t)span, -1);
      for (size_t i = 0; i < d->keys.size(); i++)
        lut[(size_t)(d->keys[i] - mn)] = d->attr_gid[i];
      d->dev_lut = (const int32_t *)up(e, lut.data(), lut.size() * 4);
      d->lut_min = mn; d->lut_max = mx;
    }
  }
  return SN_OK;
}

static Table *get_table(sn_engine *e, int32_t t) {
  if (!e || t < 0 || t >= (int32_t)e->tables.size()) return nullptr;
  return e->tables[t].get();
}

/* ---- LZ4 compression wrapper (CompressionUtils.scala:53-61,132-160):
 * [int32 -codecId][int32 uncompressedLen][payload]; codec 1 = LZ4 (default),
 * 2 = Snappy.  The engine decompresses on put, mirroring
 * ColumnFormatValue.getValueRetain(DECOMPRESS)
 * (ColumnFormatEntry.scala:267-330).  liblz4 loaded lazily via dlopen
 * (runtime-only .so in this image). ---- */
typedef int (*lz4_decomp_fn)(const char *, char *, int, int);
static lz4_decomp_fn get_lz4(void) {
  static lz4_decomp_fn fn = nullptr;
  static bool tried = false;
  if (!tried) {
    tried = true;
    void *h = dlopen("liblz4.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("liblz4.so", RTLD_NOW | RTLD_GLOBAL);
    if (h) fn = (lz4_decomp_fn)dlsym(h, "LZ4_decompress_safe");
  }
  return fn;
}

/* if blob is codec-wrapped, decompress into *out and return 1; 0 if plain;
 * negative = error */
/* Raw Snappy block decompress (format_description.txt of google/snappy —
 * the raw block format snappy-java's Snappy.compress emits, which the
 * reference wraps as codec 2, CompressionCodecId.scala:31).  No libsnappy
 * ships in this image, so the decoder is implemented here: varint32
 * uncompressed length, then literal (tag&3==0) and copy (1/2/4-byte
 * offset) elements; copies may overlap and run byte-by-byte. */
static int snappy_decompress(const uint8_t *src, int64_t slen,
                             std::vector<uint8_t> *out) {
  int64_t ip = 0;
  uint32_t ulen = 0;
  int shift = 0;
  while (ip < slen) {
    uint8_t b = src[ip++];
    ulen |= (uint32_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
    if (shift > 28) return -1;
  }
  out->resize(ulen);
  uint8_t *dst = out->data();
  uint32_t op = 0;
  while (ip < slen) {
    uint8_t tag = src[ip++];
    if ((tag & 3) == 0) {                        /* literal */
      uint32_t n = (uint32_t)(tag >> 2) + 1;
      if (n > 60) {
        int nb = (int)n - 60;
        if (ip + nb > slen) return -1;
        n = 0;
        for (int i = 0; i < nb; i++) n |= (uint32_t)src[ip + i] << (8 * i);
        n += 1;
        ip += nb;
      }
      if (ip + n > slen || (uint64_t)op + n > ulen) return -1;
      memcpy(dst + op, src + ip, n);
      ip += n; op += n;
    } else {                                     /* copy */
      uint32_t n, off;
      if ((tag & 3) == 1) {
        if (ip >= slen) return -1;
        n = ((uint32_t)(tag >> 2) & 7) + 4;
        off = ((uint32_t)(tag >> 5) << 8) | src[ip++];
      } else if ((tag & 3) == 2) {
        if (ip + 2 > slen) return -1;
        n = (uint32_t)(tag >> 2) + 1;
        off = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8);
        ip += 2;
      } else {
        if (ip + 4 > slen) return -1;
        n = (uint32_t)(tag >> 2) + 1;
        off = (uint32_t)src[ip] | ((uint32_t)src[ip + 1] << 8) |
              ((uint32_t)src[ip + 2] << 16) | ((uint32_t)src[ip + 3] << 24);
        ip += 4;
      }
      if (off == 0 || off > op || (uint64_t)op + n > ulen) return -1;
      for (uint32_t i = 0; i < n; i++) { dst[op] = dst[op - off]; op++; }
    }
  }
  return op == ulen ? 0 : -1;
}

static int maybe_decompress(const uint8_t *blob, int64_t len,
                            std::vector<uint8_t> *out) {
  if (len < 8) return 0;
  int32_t tag = rd_i32(blob);
  if (tag >= 0) return 0;                  /* plain typeId */
  int32_t codec = -tag;
  int32_t ulen = rd_i32(blob + 4);
  if (ulen <= 0) return fail(SN_ERR_BADFORMAT, "bad compressed length");
  if (codec == 1) {
    lz4_decomp_fn fn = get_lz4();
    if (!fn) return fail(SN_ERR_UNSUPPORTED, "liblz4 unavailable");
    out->resize((size_t)ulen);
    int n = fn((const char *)blob + 8, (char *)out->data(),
               (int)(len - 8), ulen);
    if (n != ulen) return fail(SN_ERR_BADFORMAT, "LZ4 decompress failed (%d)", n);
    return 1;
  }
  if (codec == 2) {
    if (snappy_decompress(blob + 8, len - 8, out) != 0 ||
        (int64_t)out->size() != ulen)
      return fail(SN_ERR_BADFORMAT, "Snappy decompress failed");
    return 1;
  }
  return fail(SN_ERR_UNSUPPORTED, "codec %d not supported", codec);
}

/* ---- blob header parse (PRODUCT-side restatement of
 *      ColumnEncoding.scala:37-53,764-832; DictionaryEncoding.scala:85-160) */
static int parse_blob(const uint8_t *blob, int64_t len, sn_type_t dtype,
                      ColMeta *m) {
  if (!blob || len < 8) return SN_ERR_BADFORMAT;
  m->type_id = rd_i32(blob);
  if (m->type_id < 0 || m->type_id > 4) return SN_ERR_BADFORMAT;
  int64_t cur = 4;
  int32_t null_bytes = rd_i32(blob + cur); cur += 4;
  if (null_bytes < 0 || (null_bytes & 7) || cur + null_bytes > len) return SN_ERR_BADFORMAT;
  if (null_bytes) {
    m->null_off = cur;
    m->num_null_words = null_bytes >> 3;
    cur += null_bytes;
  }
  if (m->type_id == SN_ENC_DICTIONARY || m->type_id == SN_ENC_BIG_DICTIONARY) {
    if (cur + 4 > len) return SN_ERR_BADFORMAT;
    int32_t n = rd_i32(blob + cur); cur += 4;
    if (n < 0) return SN_ERR_BADFORMAT;
    m->dict_n = n;
    if (dtype == SN_TYPE_STRING) {
      m->dict.reserve(n);
      for (int32_t i = 0; i < n; i++) {
        if (cur + 4 > len) return SN_ERR_BADFORMAT;
        int32_t sz = rd_i32(blob + cur); cur += 4;
        if (sz < 0 || cur + sz > len) return SN_ERR_BADFORMAT;
        m->dict.emplace_back((const char *)blob + cur, (size_t)sz);
        cur += sz;
      }
    } else if (dtype == SN_TYPE_INT32) {
      for (int32_t i = 0; i < n; i++) { m->dict_i.push_back(rd_i32(blob + cur)); cur += 4; }
    } else if (dtype == SN_TYPE_INT64) {
      for (int32_t i = 0; i < n; i++) { m->dict_i.push_back(rd_i64(blob + cur)); cur += 8; }
    } else return SN_ERR_UNSUPPORTED;
    if (cur > len) return SN_ERR_BADFORMAT;
  }
  m->body_off = cur;
  return SN_OK;
}

/* decode one delta blob into (pos, value, isnull) triples.
 * Restates ColumnDeltaDecoder.scala:31-120 on the host (deltas are bounded
 * by ColumnMaxDeltaRows, so this is O(10^4) host work per column). */
static int decode_delta(const uint8_t *blob, int64_t len, sn_type_t dtype,
                        Table *tab, int col,
                        std::vector<int32_t> *pos, std::vector<double> *val,
                        std::vector<uint8_t> *isnull) {
  ColMeta m;
  /* header first: typeId + nulls (over delta ENTRIES) */
  if (!blob || len < 16) return SN_ERR_BADFORMAT;
  int type_id = rd_i32(blob);
  int64_t cur = 4;
  int32_t null_bytes = rd_i32(blob + cur); cur += 4;
  if (null_bytes < 0 || (null_bytes & 7)) return SN_ERR_BADFORMAT;
  const uint8_t *nullw = null_bytes ? blob + cur : nullptr;
  cur += null_bytes;
  /* positions section (ColumnDeltaDecoder.initialize :45-58) */
  int32_t npos = rd_i32(blob + cur + 4);
  if (npos < 0) return SN_ERR_BADFORMAT;
  const uint8_t *posp = blob + cur + 8;
  cur = cur + 8 + (int64_t)npos * 4;
  cur = (cur + 7) & ~7ll;
  if (cur > len) return SN_ERR_BADFORMAT;
  /* body (encoding-specific) */
  m.type_id = type_id;
  int32_t dict_n = 0;
  std::vector<std::string> dict;
  std::vector<int64_t> dict_i;
  if (type_id == SN_ENC_DICTIONARY || type_id == SN_ENC_BIG_DICTIONARY) {
    dict_n = rd_i32(blob + cur); cur += 4;
    if (dtype == SN_TYPE_STRING) {
      for (int32_t i = 0; i < dict_n; i++) {
        int32_t sz = rd_i32(blob + cur); cur += 4;
        dict.emplace_back((const char *)blob + cur, (size_t)sz); cur += sz;
      }
    } else {
      int w = dtype == SN_TYPE_INT64 ? 8 : 4;
      for (int32_t i = 0; i < dict_n; i++) {
        dict_i.push_back(w == 8 ? rd_i64(blob + cur) : rd_i32(blob + cur)); cur += w;
      }
    }
  } else if (type_id != SN_ENC_UNCOMPRESSED) {
    return SN_ERR_UNSUPPORTED;  /* RLE deltas not produced by this build */
  }
  const uint8_t *body = blob + cur;
  int nnp = 0;
  int64_t var_cur = 0;
  for (int32_t i = 0; i < npos; i++) {
    int32_t p = rd_i32(posp + (int64_t)i * 4);
    bool nul = nullw && ((rd_i64(nullw + ((i >> 6) << 3)) >> (i & 63)) & 1);
    pos->push_back(p);
    isnull->push_back(nul ? 1 : 0);
    if (nul) { val->push_back(0.0); continue; }
    double v = 0.0;
    if (type_id == SN_ENC_UNCOMPRESSED) {
      switch (dtype) {
        case SN_TYPE_DOUBLE: v = rd_f64(body + (int64_t)nnp * 8); break;
        case SN_TYPE_FLOAT:  v = rd_f32(body + (int64_t)nnp * 4); break;
        case SN_TYPE_INT32:  v = (double)rd_i32(body + (int64_t)nnp * 4); break;
        case SN_TYPE_INT64: {
          /* carry RAW BITS (exact beyond 2^53, like the base-column path);
           * every consumer of int64 patch values bitcasts back */
          int64_t iv = rd_i64(body + (int64_t)nnp * 8);
          memcpy(&v, &iv, 8);
          break;
        }
        case SN_TYPE_INT16:  v = (double)rd_i16(body + (int64_t)nnp * 2); break;
        case SN_TYPE_STRING: {
          int32_t sz = rd_i32(body + var_cur);
          std::string s((const char *)body + var_cur + 4, (size_t)sz);
          var_cur += 4 + sz;
          /* intern into the table-global dictionary */
          auto &gi = tab->gdict_idx[col];
          auto it = gi.find(s);
          int32_t gid;
          if (it == gi.end()) {
            gid = (int32_t)tab->gdict[col].size();
            tab->gdict[col].push_back(s);
            gi.emplace(s, gid);
            if ((int32_t)s.size() > tab->gdict_maxlen[col])
              tab->gdict_maxlen[col] = (int32_t)s.size();
          } else gid = it->second;
          v = (double)gid;
          break;
        }
        default: return SN_ERR_UNSUPPORTED;
      }
    } else { /* dictionary-encoded delta body */
      int32_t idx = m.type_id == SN_ENC_DICTIONARY
          ? (int32_t)(uint16_t)rd_i16(body + (int64_t)nnp * 2)
          : rd_i32(body + (int64_t)nnp * 4);
      if (dtype == SN_TYPE_STRING) {
        if (idx >= dict_n) { isnull->back() = 1; val->push_back(0.0); nnp++; continue; }
        auto &gi = tab->gdict_idx[col];
        const std::string &s = dict[idx];
        auto it = gi.find(s);
        int32_t gid;
        if (it == gi.end()) {
          gid = (int32_t)tab->gdict[col].size();
          tab->gdict[col].push_back(s);
          gi.emplace(s, gid);
          if ((int32_t)s.size() > tab->gdict_maxlen[col])
            tab->gdict_maxlen[col] = (int32_t)s.size();
        } else gid = it->second;
        v = (double)gid;
      } else if (dtype == SN_TYPE_INT64) {
        int64_t iv = dict_i[idx];
        memcpy(&v, &iv, 8);              /* raw bits, same as above */
      } else {
        v = (double)dict_i[idx];
      }
    }
    val->push_back(v);
    nnp++;
  }
  return SN_OK;
}

/* upload helper */
static void *up(sn_engine *e, const void *host, size_t n) {
  void *d = e->arena.alloc(n);
  if (!d) return nullptr;
  if (e->arena.device >= 0) {
    if (hipMemcpy(d, host, n, hipMemcpyHostToDevice) != hipSuccess) return nullptr;
  } else {
    memcpy(d, host, n);
  }
  return d;
}

/* ---- batch mutation processing (shared by sn_batch_put and
 * sn_batch_mutate — the reference writes delete masks and delta blobs to
 * region entries of an EXISTING batch after UPDATE/DELETE statements;
 * the ColumnBatchIterator hands the current state per scan) ---- */

static int32_t apply_delete_mask(sn_engine *e, Batch &b,
                                 const sn_buf *delete_mask) {
  /* delete mask -> bitmap (ColumnDeleteDecoder.scala:24-55 semantics);
   * the mask is cumulative, so it REPLACES any previous one */
  b.has_deletes = false;
  b.del_bm_dev = nullptr;
  if (delete_mask && delete_mask->data && delete_mask->len < 12)
    return fail(SN_ERR_BADFORMAT, "delete mask shorter than its header");
  if (delete_mask && delete_mask->data && delete_mask->len >= 12) {
    const uint8_t *dm = (const uint8_t *)delete_mask->data;
    int32_t n = rd_i32(dm + 8);
    if (n < 0 || 12 + (int64_t)n * 4 > delete_mask->len)
      return fail(SN_ERR_BADFORMAT, "bad delete mask");
    if (n > 0) {
      b.has_deletes = true;
      std::vector<uint64_t> bm(((size_t)b.num_rows + 63) / 64, 0);
      for (int32_t i = 0; i < n; i++) {
        int32_t p = rd_i32(dm + 12 + (int64_t)i * 4);
        if (p >= 0 && p < b.num_rows) bm[p >> 6] |= 1ull << (p & 63);
      }
      b.del_bm_dev = (const uint64_t *)up(e, bm.data(), bm.size() * 8);
    }
  }

  return SN_OK;
}

static int32_t apply_deltas(sn_engine *e, Table *t, Batch &b,
                            const sn_buf *deltas) {
  /* update deltas -> host-merged patches (delta1 overrides delta2,
   * UpdatedColumnDecoder.scala:69-115).  Delta blobs are cumulative for
   * their batch (the encoder merges upward), so the decoded set REPLACES
   * any previous patch state; values already materialized into the body
   * are simply rewritten with the same bytes. */
  const int nc = (int)t->schema.size();
  if (deltas) {
    for (int c = 0; c < nc; c++) {
      const sn_buf &d1 = deltas[c * 2], &d2 = deltas[c * 2 + 1];
      if (!d1.data && !d2.data) continue;
      /* deltas may arrive with a positive row count too (the reference
       * signals them via the stats row's negative batchCount); either way
       * the batch must never stats-skip — base bounds don't cover patches */
      b.has_deltas = true;
      b.patch_host[c] = Patch();          /* cumulative: replace prior state */
      b.patch_dev[c] = Batch::PatchDev();
      std::vector<int32_t> p1, p2;
      std::vector<double> v1, v2;
      std::vector<uint8_t> n1, n2;
      if (d2.data) {
        int rc = decode_delta((const uint8_t *)d2.data, d2.len, t->schema[c].dtype,
                              t, c, &p2, &v2, &n2);
        if (rc != SN_OK) return fail(rc, "delta2 decode col %d", c);
      }
      if (d1.data) {
        int rc = decode_delta((const uint8_t *)d1.data, d1.len, t->schema[c].dtype,
                              t, c, &p1, &v1, &n1);
        if (rc != SN_OK) return fail(rc, "delta1 decode col %d", c);
      }
      /* merge: start from delta2, override with delta1 */
      std::map<int32_t, std::pair<double, uint8_t>> merged;
      for (size_t i = 0; i < p2.size(); i++) merged[p2[i]] = { v2[i], n2[i] };
      for (size_t i = 0; i < p1.size(); i++) merged[p1[i]] = { v1[i], n1[i] };
      Patch &P = b.patch_host[c];
      for (auto &kv : merged) {
        P.pos.push_back(kv.first);
        P.val.push_back(kv.second.first);
        P.isnull.push_back(kv.second.second);
      }
      if (P.pos.empty()) continue;
      b.had_patches[c] = 1;
      /* device structures */
      std::vector<uint64_t> bm(((size_t)b.num_rows + 63) / 64, 0);
      for (int32_t p : P.pos) if (p >= 0 && p < b.num_rows) bm[p >> 6] |= 1ull << (p & 63);
      std::vector<uint64_t> nbm((P.pos.size() + 63) / 64, 0);
      bool any_null = false;
      for (size_t i = 0; i < P.isnull.size(); i++)
        if (P.isnull[i]) { nbm[i >> 6] |= 1ull << (i & 63); any_null = true; }
      auto &pd = b.patch_dev[c];
      pd.n = (int32_t)P.pos.size();
      pd.bm = (const uint64_t *)up(e, bm.data(), bm.size() * 8);
      pd.pos = (const int32_t *)up(e, P.pos.data(), P.pos.size() * 4);
      pd.val = (const double *)up(e, P.val.data(), P.val.size() * 8);
      pd.nullbm = any_null ? (const uint64_t *)up(e, nbm.data(), nbm.size() * 8) : nullptr;

      /* materialize value-only patches straight into the device body when
       * the base column is null-free fixed-width: the batch then scans
       * clean (query-compiled kernels apply).  Scan results are identical
       * by construction — read_general would hand back exactly these
       * values; the host blob (sn_table_get_blob) keeps the base bytes. */
      if (e->has_gpu && !any_null && b.cols[c].num_null_words == 0 &&
          b.cols[c].type_id == SN_ENC_UNCOMPRESSED) {
        int k = -1;
        switch (t->schema[c].dtype) {
          case SN_TYPE_DOUBLE: k = SN_K_F64; break;
          case SN_TYPE_FLOAT:  k = SN_K_F32; break;
          case SN_TYPE_INT32:  k = SN_K_I32; break;
          case SN_TYPE_INT64:  k = SN_K_I64; break;
          case SN_TYPE_INT16:  k = SN_K_I16; break;
          default: break;
        }
        if (k >= 0) {
          void *body = (uint8_t *)(uintptr_t)b.col_dev[c] + b.cols[c].body_off;
          if (sn_launch_patch_apply(body, pd.pos, pd.val, pd.n, k,
                                    e->stream) == 0) {
            pd = Batch::PatchDev();
            P = Patch();
          }
        }
      }
    }
  }

  return SN_OK;
}

static void apply_stats(Table *t, Batch &b, const sn_buf *stats) {
  /* stats row parse (UnsafeRow: [null words][3*ncols+1 x 8B slots],
   * ColumnStatsSchema, ColumnEncoding.scala:1015-1036) */
  const int nc = (int)t->schema.size();
  if (stats && stats->data) {
    const uint8_t *sp = (const uint8_t *)stats->data;
    int32_t num_fields = nc * 3 + 1;
    int32_t nwords = (num_fields + 63) >> 6;
    if (stats->len >= (int64_t)nwords * 8 + (int64_t)num_fields * 8) {
      const uint8_t *bits = sp;
      const uint8_t *slots = sp + (int64_t)nwords * 8;
      auto bit = [&](int f) {
        return (rd_i64(bits + ((f >> 6) << 3)) >> (f & 63)) & 1;
      };
      b.lo_d.resize(nc); b.hi_d.resize(nc);
      b.lo_i.resize(nc); b.hi_i.resize(nc);
      b.null_count.resize(nc); b.bounds_null.resize(nc);
      for (int c = 0; c < nc; c++) {
        int f_lo = 1 + c * 3, f_hi = 2 + c * 3, f_nc = 3 + c * 3;
        b.bounds_null[c] = bit(f_lo) || bit(f_hi);
        b.null_count[c] = bit(f_nc) ? 0 : rd_i32(slots + (int64_t)f_nc * 8);
        if (b.bounds_null[c]) continue;
        switch (t->schema[c].dtype) {
          case SN_TYPE_DOUBLE:
            b.lo_d[c] = rd_f64(slots + (int64_t)f_lo * 8);
            b.hi_d[c] = rd_f64(slots + (int64_t)f_hi * 8); break;
          case SN_TYPE_FLOAT:
            b.lo_d[c] = rd_f32(slots + (int64_t)f_lo * 8);
            b.hi_d[c] = rd_f32(slots + (int64_t)f_hi * 8); break;
          case SN_TYPE_INT64:
            b.lo_i[c] = rd_i64(slots + (int64_t)f_lo * 8);
            b.hi_i[c] = rd_i64(slots + (int64_t)f_hi * 8); break;
          case SN_TYPE_STRING:
            b.bounds_null[c] = 1; break;
          default:
            b.lo_i[c] = rd_i32(slots + (int64_t)f_lo * 8);
            b.hi_i[c] = rd_i32(slots + (int64_t)f_hi * 8);
        }
      }
      b.stats_valid = true;
    }
  }

}

/* one encoded column blob -> device-resident state (decompress, parse,
 * transcode/materialize, validate, upload, aux) — batch-local, no table
 * lock; shared by sn_batch_put and sn_batch_put_raw */
static int32_t process_encoded_col(sn_engine *e, Table *t, Batch &b, int c,
                                   const sn_buf &column) {
  std::vector<uint8_t> decomp;
    const uint8_t *blob = (const uint8_t *)column.data;
    int64_t len = column.len;
    int dec = maybe_decompress(blob, len, &decomp);
    if (dec < 0) return dec;
    if (dec == 1) { blob = decomp.data(); len = (int64_t)decomp.size(); }
    int rc = parse_blob(blob, len, t->schema[c].dtype, &b.cols[c]);
    if (rc != SN_OK) return fail(rc, "column %d blob parse failed", c);
    if (t->schema[c].dtype == SN_TYPE_STRING && !b.cols[c].dict.empty()) {
      /* validate the index array: a corrupt index would walk the kernel
       * off the end of the local->global map (index == dict size is the
       * NULL sentinel, DictionaryEncoding.scala:90) */
      const ColMeta &dm = b.cols[c];
      const int32_t dn = (int32_t)dm.dict.size();
      const uint8_t *bodyp = blob + dm.body_off;
      int w = dm.type_id == SN_ENC_DICTIONARY ? 2 : 4;
      /* exactly one index per NON-NULL row; padding beyond is not data */
      int64_t nnull = 0;
      for (int32_t wi = 0; wi < dm.num_null_words; wi++)
        nnull += __builtin_popcountll(
            (unsigned long long)rd_i64(blob + dm.null_off + (int64_t)wi * 8));
      int64_t nidx = (int64_t)b.num_rows - nnull;
      if (nidx * w > len - dm.body_off)
        return fail(SN_ERR_BADFORMAT, "dictionary index array truncated col %d", c);
      for (int64_t i = 0; i < nidx; i++) {
        int32_t ix = w == 2 ? (int32_t)(uint16_t)rd_i16(bodyp + i * 2)
                            : rd_i32(bodyp + i * 4);
        if (ix < 0 || ix > dn)
          return fail(SN_ERR_BADFORMAT,
                      "dictionary index %d out of range [0,%d] at row %lld col %d",
                      ix, dn, (long long)i, c);
      }
    }
    /* Var-width (non-dictionary) STRING bodies (Uncompressed.scala:117-160:
     * [int32 size][bytes] sequential per non-null row) have no random-access
     * layout a data-parallel kernel can consume; transcode them to
     * Dictionary encoding at put (the same re-encode-on-put idea as the
     * LZ4 wrapper and int-dict materialization below) so the existing
     * dictionary machinery — global interning, premultiplied maps, pushdown
     * equality, group keys — applies wholesale. */
    std::vector<uint8_t> synth;
    if (t->schema[c].dtype == SN_TYPE_STRING &&
        b.cols[c].type_id == SN_ENC_UNCOMPRESSED) {
      const ColMeta dm = b.cols[c];
      const int32_t rows = b.num_rows;
      const int32_t nwords = dm.num_null_words;
      std::map<std::string, int32_t> didx;
      std::vector<std::string> dict;
      std::vector<int32_t> idx;
      idx.reserve(rows);
      int64_t cur = dm.body_off;
      int64_t nnull = 0;
      for (int32_t wi = 0; wi < nwords; wi++)
        nnull += __builtin_popcountll(
            (unsigned long long)rd_i64(blob + dm.null_off + (int64_t)wi * 8));
      const int64_t nn_rows = rows - nnull;
      int64_t dict_bytes = 0;
      for (int64_t i = 0; i < nn_rows; i++) {
        if (cur + 4 > len)
          return fail(SN_ERR_BADFORMAT, "string body truncated col %d", c);
        int32_t sz = rd_i32(blob + cur);
        cur += 4;
        if (sz < 0 || cur + sz > len)
          return fail(SN_ERR_BADFORMAT, "string body truncated col %d", c);
        std::string s((const char *)blob + cur, (size_t)sz);
        cur += sz;
        auto it = didx.find(s);
        int32_t di;
        if (it == didx.end()) {
          di = (int32_t)dict.size();
          dict_bytes += 4 + sz;
          didx.emplace(std::move(s), di);
          dict.push_back(std::string((const char *)blob + cur - sz, (size_t)sz));
        } else di = it->second;
        idx.push_back(di);
      }
      const int64_t nb = (int64_t)nwords * 8;
      synth.resize(8 + nb + 4 + dict_bytes + (int64_t)idx.size() * 4);
      int32_t tid3 = SN_ENC_BIG_DICTIONARY, nb32 = (int32_t)nb;
      memcpy(synth.data(), &tid3, 4);
      memcpy(synth.data() + 4, &nb32, 4);
      if (nb) memcpy(synth.data() + 8, blob + dm.null_off, (size_t)nb);
      uint8_t *w = synth.data() + 8 + nb;
      int32_t dn = (int32_t)dict.size();
      memcpy(w, &dn, 4); w += 4;
      for (auto &s : dict) {
        int32_t sz = (int32_t)s.size();
        memcpy(w, &sz, 4); w += 4;
        memcpy(w, s.data(), s.size()); w += s.size();
      }
      memcpy(w, idx.data(), idx.size() * 4);
      blob = synth.data();
      len = (int64_t)synth.size();
      b.cols[c] = ColMeta();
      int rc2 = parse_blob(blob, len, SN_TYPE_STRING, &b.cols[c]);
      if (rc2 != SN_OK) return fail(rc2, "transcoded string col %d", c);
    }

    /* int-typed dictionary columns (DictionaryEncoding over int32/int64,
     * DictionaryEncoding.scala:85-137): materialize the values into a plain
     * fixed-width body at put — decompress-on-put, like the LZ4 wrapper —
     * so the scan runs the plain I32/I64 path (JIT-eligible).  Index ==
     * numElements is the null sentinel (DictionaryEncoding.scala:90): such
     * rows are folded into the synthesized null bitset. */
    if ((b.cols[c].type_id == SN_ENC_DICTIONARY ||
         b.cols[c].type_id == SN_ENC_BIG_DICTIONARY) &&
        t->schema[c].dtype != SN_TYPE_STRING) {
      const ColMeta dm = b.cols[c];
      const sn_type_t dt = t->schema[c].dtype;
      const int w = dm.type_id == SN_ENC_DICTIONARY ? 2 : 4;
      const int vw = dt == SN_TYPE_INT64 ? 8 : 4;
      const int32_t dn = dm.dict_n;
      const int32_t rows = b.num_rows;
      const int32_t nwords = (rows + 63) >> 6;
      std::vector<uint64_t> nullw(nwords, 0);
      if (dm.num_null_words) {
        for (int32_t wi = 0; wi < dm.num_null_words && wi < nwords; wi++)
          nullw[wi] = (uint64_t)rd_i64(blob + dm.null_off + (int64_t)wi * 8);
      }
      std::vector<int64_t> vals;
      vals.reserve(rows);
      const uint8_t *bodyp = blob + dm.body_off;
      int64_t ii = 0;                     /* index cursor (one per non-null row) */
      const int64_t navail = (len - dm.body_off) / w;
      for (int32_t r = 0; r < rows; r++) {
        if ((nullw[r >> 6] >> (r & 63)) & 1) continue;
        if (ii >= navail)
          return fail(SN_ERR_BADFORMAT, "dictionary index array truncated col %d", c);
        int32_t ix = w == 2 ? (int32_t)(uint16_t)rd_i16(bodyp + ii * 2)
                            : rd_i32(bodyp + ii * 4);
        ii++;
        if (ix < 0 || ix > dn)
          return fail(SN_ERR_BADFORMAT,
                      "dictionary index %d out of range [0,%d] col %d", ix, dn, c);
        if (ix == dn) { nullw[r >> 6] |= 1ull << (r & 63); continue; }
        vals.push_back(dm.dict_i[ix]);
      }
      bool any_null = false;
      for (uint64_t wv : nullw) any_null |= wv != 0;
      const int64_t nb = any_null ? (int64_t)nwords * 8 : 0;
      synth.resize(8 + nb + (int64_t)vals.size() * vw);
      int32_t tid0 = SN_ENC_UNCOMPRESSED, nb32 = (int32_t)nb;
      memcpy(synth.data(), &tid0, 4);
      memcpy(synth.data() + 4, &nb32, 4);
      if (nb) memcpy(synth.data() + 8, nullw.data(), (size_t)nb);
      uint8_t *vb = synth.data() + 8 + nb;
      for (size_t i = 0; i < vals.size(); i++) {
        if (vw == 8) memcpy(vb + i * 8, &vals[i], 8);
        else { int32_t v32 = (int32_t)vals[i]; memcpy(vb + i * 4, &v32, 4); }
      }
      blob = synth.data();
      len = (int64_t)synth.size();
      b.cols[c] = ColMeta();
      int rc2 = parse_blob(blob, len, dt, &b.cols[c]);
      if (rc2 != SN_OK) return fail(rc2, "materialized dict col %d", c);
    }

    /* upload blob, placed so the BODY is 16-byte aligned (the scan kernel
     * issues 16 B/lane vector loads on the body; the 8-byte blob header
     * would otherwise leave it 8-aligned).
     * f1 compressed-upload: LZ4-wrapped blobs ship their COMPRESSED bytes
     * over PCIe (the host decompressed copy serves only parse/validation)
     * and decode wave-cooperatively on device into the blob slot —
     * byte-identical output, ~2-4x less bus traffic per blob. */
    {
      int64_t pad = (16 - (b.cols[c].body_off & 15)) & 15;
      char *base = (char *)e->arena.alloc((size_t)len + 16);
      if (!base) return fail(SN_ERR_NOMEM, "HBM upload failed");
      char *dst = base + pad;
      bool dev_decoded = false;
      if (e->arena.device >= 0 && dec == 1 &&
          rd_i32((const uint8_t *)column.data) == -1 /* LZ4 */) {
        const int64_t clen = column.len - 8;
        /* the engine stream + error word are shared: one decode at a time
         * (concurrent puts overlap their host work; device decodes queue) */
        std::lock_guard<std::mutex> glz(e->lz4_mu);
        void *cdev = e->arena.alloc((size_t)clen);
        if (!e->lz4_err_dev)
          e->lz4_err_dev = (int32_t *)e->arena.alloc(64);
        if (cdev && e->lz4_err_dev &&
            hipMemcpy(cdev, (const uint8_t *)column.data + 8,
                      (size_t)clen, hipMemcpyHostToDevice) == hipSuccess &&
            hipMemsetAsync(e->lz4_err_dev, 0, 4, e->stream) == hipSuccess &&
            sn_launch_lz4_decompress(cdev, clen, dst, len, e->lz4_err_dev,
                                     e->stream) == 0 &&
            hipStreamSynchronize(e->stream) == hipSuccess) {
          int32_t derr = -1;
          (void)hipMemcpy(&derr, e->lz4_err_dev, 4, hipMemcpyDeviceToHost);
          if (derr != 0)
            return fail(SN_ERR_BADFORMAT,
                        "device LZ4 decode failed (%d) col %d", derr, c);
          dev_decoded = true;
        }
        if (cdev) e->arena.release(cdev, (size_t)clen);
      }
      if (!dev_decoded) {
        if (e->arena.device >= 0) {
          if (h2d_copy(e, dst, blob, (size_t)len) != hipSuccess)
            return fail(SN_ERR_NOMEM, "HBM upload failed");
        } else {
          memcpy(dst, blob, (size_t)len);
        }
      }
      b.col_dev[c] = dst;
    }
    if (e->arena.device < 0) {
      b.host_blobs.emplace_back(blob, blob + len);
    }
    /* RLE aux: extract (cumulative end, value-as-f64) per run on host
     * (the blob stores [value][int32 run-length] entries; the decoder
     * accumulates — RunLengthEncoding.scala:99-172) */
    if (b.cols[c].type_id == SN_ENC_RUNLENGTH) {
      sn_type_t dt = t->schema[c].dtype;
      int w = dt == SN_TYPE_INT16 ? 2 : dt == SN_TYPE_INT32 ? 4 :
              dt == SN_TYPE_INT64 ? 8 : 0;
      if (w == 0)
        return fail(SN_ERR_UNSUPPORTED, "RLE col %d dtype %d on GPU path", c, dt);
      std::vector<int32_t> ends;
      std::vector<double> vals;
      int64_t cur = b.cols[c].body_off;
      int32_t acc = 0;
      while (cur + w + 4 <= len) {
        int64_t v = w == 2 ? rd_i16(blob + cur) : w == 4 ? rd_i32(blob + cur)
                                                         : rd_i64(blob + cur);
        acc += rd_i32(blob + cur + w);
        ends.push_back(acc);
        if (dt == SN_TYPE_INT64) {
          double d; memcpy(&d, &v, 8); vals.push_back(d);  /* raw bits */
        } else vals.push_back((double)v);
        cur += w + 4;
      }
      if (!ends.empty()) {
        b.rle_ends_dev[c] = (const int32_t *)up(e, ends.data(), ends.size() * 4);
        b.rle_vals_dev[c] = (const double *)up(e, vals.data(), vals.size() * 8);
        b.rle_n[c] = (int32_t)ends.size();
      }
    }

    /* null prefix aux */
    if (b.cols[c].num_null_words) {
      const uint8_t *nw = blob + b.cols[c].null_off;
      int W = b.cols[c].num_null_words;
      std::vector<uint32_t> pfx((size_t)W);
      uint32_t acc = 0;
      for (int w = 0; w < W; w++) {
        pfx[w] = acc;
        acc += (uint32_t)__builtin_popcountll((unsigned long long)rd_i64(nw + (int64_t)w * 8));
      }
      b.nullpfx_dev[c] = (const uint32_t *)up(e, pfx.data(), pfx.size() * 4);
    }
    return SN_OK;
}

/* f2 fast ingest: fixed-width NON-NULL columns arrive as RAW value arrays
 * (the Uncompressed encoder for numerics is the identity), other columns as
 * encoded blobs.  Bounds for the raw columns compute ON DEVICE (async
 * min/max on the engine stream — ColumnEncoder's lowerLong/upperLong
 * tracking, ColumnEncoding.scala:188-251, moved to the GPU); queries force
 * the pending sync.  This is the ColumnBatchCreator rollover with encode +
 * stats offloaded — ingest threads only touch the bytes once. */
extern "C" int32_t sn_batch_put_raw(sn_engine *e, int32_t table,
                                    int64_t uuid, int32_t bucket_id,
                                    int32_t num_rows, const sn_buf *raw,
                                    const sn_buf *encoded) {
  Table *t = get_table(e, table);
  if (!t || !raw || !encoded || num_rows <= 0)
    return fail(SN_ERR_BADARG, "bad raw batch args");
  if (e->cfg.shard_count > 1 &&
      (bucket_id % e->cfg.shard_count) != e->cfg.shard_rank)
    return SN_OK;
  const int nc = (int)t->schema.size();
  Batch b;
  b.uuid = uuid;
  b.bucket = bucket_id;
  b.num_rows = num_rows;
  b.cols.resize(nc);
  b.col_dev.resize(nc);
  b.nullpfx_dev.resize(nc, nullptr);
  b.patch_dev.resize(nc);
  b.patch_host.resize(nc);
  b.had_patches.assign(nc, 0);
  b.rle_ends_dev.resize(nc, nullptr);
  b.rle_vals_dev.resize(nc, nullptr);
  b.rle_n.resize(nc, 0);
  b.dictmap_cache.resize(nc);
  b.is_raw.assign(nc, 0);

  unsigned long long *sd = nullptr;
  if (e->has_gpu) {
    sd = (unsigned long long *)e->arena.alloc((size_t)nc * 16);
    if (!sd) return fail(SN_ERR_NOMEM, "stats scratch");
    if (hipMemsetAsync(sd, 0xff, (size_t)nc * 8, e->stream) != hipSuccess ||
        hipMemsetAsync(sd + nc, 0, (size_t)nc * 8, e->stream) != hipSuccess)
      return fail(SN_ERR_GENERIC, "stats scratch init");
  }
  bool any_raw = false;
  for (int c = 0; c < nc; c++) {
    if (!raw[c].data) {
      if (!encoded[c].data)
        return fail(SN_ERR_BADARG, "column %d has neither raw nor encoded", c);
      int32_t rc_pc = process_encoded_col(e, t, b, c, encoded[c]);
      if (rc_pc != SN_OK) return rc_pc;
      continue;
    }
    sn_type_t dt = t->schema[c].dtype;
    int w, kind;
    switch (dt) {
      case SN_TYPE_DOUBLE: w = 8; kind = SN_K_F64; break;
      case SN_TYPE_INT64:  w = 8; kind = SN_K_I64; break;
      case SN_TYPE_INT32:  w = 4; kind = SN_K_I32; break;
      case SN_TYPE_FLOAT:  w = 4; kind = SN_K_F32; break;
      case SN_TYPE_INT16:  w = 2; kind = SN_K_I16; break;
      default:
        return fail(SN_ERR_UNSUPPORTED, "raw column %d dtype %d", c, (int)dt);
    }
    if (raw[c].len != (int64_t)num_rows * w)
      return fail(SN_ERR_BADARG, "raw column %d length", c);
    /* device blob = [typeId=0][nullBytes=0][body]: header at base+8
     * (8-aligned), body at base+16 (16-aligned — arena blocks are
     * 256-aligned); blob pointer = base+8, body_off = 8 */
    char *base = (char *)e->arena.alloc((size_t)raw[c].len + 16);
    if (!base) return fail(SN_ERR_NOMEM, "HBM upload failed");
    char *body = base + 16;
    b.cols[c].type_id = SN_ENC_UNCOMPRESSED;
    if (e->arena.device >= 0) {
      int32_t hdr[2] = { SN_ENC_UNCOMPRESSED, 0 };
      if (hipMemcpy(base + 8, hdr, 8, hipMemcpyHostToDevice) != hipSuccess ||
          h2d_copy(e, body, raw[c].data, (size_t)raw[c].len) != hipSuccess)
        return fail(SN_ERR_NOMEM, "HBM upload failed");
      if (sn_launch_col_minmax(body, num_rows, kind, sd + c, sd + nc + c,
                               e->stream) != 0)
        return fail(SN_ERR_GENERIC, "stats kernel");
      b.is_raw[c] = 1;
      any_raw = true;
    } else {
      int32_t hdr[2] = { SN_ENC_UNCOMPRESSED, 0 };
      memcpy(base + 8, hdr, 8);
      memcpy(body, raw[c].data, (size_t)raw[c].len);
      b.host_blobs.emplace_back((const uint8_t *)base + 8,
                                (const uint8_t *)body + raw[c].len);
      /* host-only: cheap scalar bounds so stats-skip still works */
      b.lo_i.resize(nc); b.hi_i.resize(nc);
      b.lo_d.resize(nc); b.hi_d.resize(nc);
      b.null_count.resize(nc, 0);
      b.bounds_null.resize(nc, 1);
      const uint8_t *p8 = (const uint8_t *)raw[c].data;
      if (dt == SN_TYPE_DOUBLE || dt == SN_TYPE_FLOAT) {
        double mn = 0, mx = 0;
        for (int32_t i = 0; i < num_rows; i++) {
          double v = dt == SN_TYPE_DOUBLE ? rd_f64(p8 + (int64_t)i * 8)
                                          : rd_f32(p8 + (int64_t)i * 4);
          if (i == 0 || v < mn) mn = v;
          if (i == 0 || v > mx) mx = v;
        }
        b.lo_d[c] = mn; b.hi_d[c] = mx;
      } else {
        int64_t mn = 0, mx = 0;
        for (int32_t i = 0; i < num_rows; i++) {
          int64_t v = w == 8 ? rd_i64(p8 + (int64_t)i * 8)
                    : w == 4 ? rd_i32(p8 + (int64_t)i * 4)
                             : rd_i16(p8 + (int64_t)i * 2);
          if (i == 0 || v < mn) mn = v;
          if (i == 0 || v > mx) mx = v;
        }
        b.lo_i[c] = mn; b.hi_i[c] = mx;
      }
      b.bounds_null[c] = 0;
      b.stats_valid = true;
    }
    b.col_dev[c] = base + 8;
    b.cols[c].body_off = 8;
  }
  if (any_raw) {
    b.stats_dev = sd;
    b.stats_pending = true;
  } else if (sd) {
    e->arena.release(sd, (size_t)nc * 16);
  }

  std::lock_guard<std::mutex> g(t->mu);
  for (int c = 0; c < nc; c++) {
    if (t->schema[c].dtype != SN_TYPE_STRING || b.cols[c].dict.empty())
      continue;
    auto &l2g = b.cols[c].local2global;
    l2g.reserve(b.cols[c].dict.size());
    for (auto &sstr : b.cols[c].dict) {
      auto it = t->gdict_idx[c].find(sstr);
      int32_t gid;
      if (it == t->gdict_idx[c].end()) {
        gid = (int32_t)t->gdict[c].size();
        t->gdict[c].push_back(sstr);
        t->gdict_idx[c].emplace(sstr, gid);
        if ((int32_t)sstr.size() > t->gdict_maxlen[c])
          t->gdict_maxlen[c] = (int32_t)sstr.size();
      } else gid = it->second;
      l2g.push_back(gid);
    }
  }
  t->total_rows += b.num_rows;
  t->batches.push_back(std::move(b));
  return SN_OK;
}

/* resolve device-computed bounds for raw-ingested batches (one stream sync
 * for ALL pending batches; called under t->mu before any bounds are read) */
static void sync_pending_stats(sn_engine *e, Table *t) {
  bool any = false;
  for (auto &b : t->batches) any |= b.stats_pending;
  if (!any) return;
  (void)hipStreamSynchronize(e->stream);
  const int nc = (int)t->schema.size();
  std::vector<unsigned long long> h((size_t)nc * 2);
  for (auto &b : t->batches) {
    if (!b.stats_pending) continue;
    b.stats_pending = false;
    if (hipMemcpy(h.data(), b.stats_dev, (size_t)nc * 16,
                  hipMemcpyDeviceToHost) != hipSuccess)
      continue;                         /* bounds stay absent: conservative */
    b.lo_d.resize(nc); b.hi_d.resize(nc);
    b.lo_i.resize(nc); b.hi_i.resize(nc);
    b.null_count.assign(nc, 0);
    b.bounds_null.assign(nc, 1);
    for (int c = 0; c < nc; c++) {
      if (!b.is_raw[c]) continue;
      sn_type_t dt = t->schema[c].dtype;
      if (dt == SN_TYPE_DOUBLE || dt == SN_TYPE_FLOAT) {
        b.lo_d[c] = sn_ord_f64_h(h[c]);
        b.hi_d[c] = sn_ord_f64_h(h[nc + c]);
      } else {
        b.lo_i[c] = (int64_t)(h[c] ^ 0x8000000000000000ull);
        b.hi_i[c] = (int64_t)(h[nc + c] ^ 0x8000000000000000ull);
      }
      b.bounds_null[c] = 0;
    }
    b.stats_valid = true;
    e->arena.release((void *)b.stats_dev, (size_t)nc * 16);
    b.stats_dev = nullptr;
  }
}

extern "C" int32_t sn_batch_put(sn_engine *e, int32_t table,
                                int64_t uuid, int32_t bucket_id, int32_t num_rows,
                                const sn_buf *columns, const sn_buf *stats,
                                const sn_buf *delete_mask, const sn_buf *deltas) {
  Table *t = get_table(e, table);
  if (!t || !columns || num_rows == 0) return fail(SN_ERR_BADARG, "bad batch args");
  /* shard filter: batches whose bucket belongs to another rank are ignored */
  if (e->cfg.shard_count > 1 &&
      (bucket_id % e->cfg.shard_count) != e->cfg.shard_rank)
    return SN_OK;

  const int nc = (int)t->schema.size();
  Batch b;
  b.uuid = uuid; b.bucket = bucket_id;
  b.has_deltas = num_rows < 0;
  b.num_rows = num_rows < 0 ? -num_rows : num_rows;
  b.cols.resize(nc);
  b.col_dev.resize(nc);
  b.nullpfx_dev.resize(nc, nullptr);
  b.patch_dev.resize(nc);
  b.patch_host.resize(nc);
  b.had_patches.assign(nc, 0);
  b.rle_ends_dev.resize(nc, nullptr);
  b.rle_vals_dev.resize(nc, nullptr);
  b.rle_n.resize(nc, 0);
  b.dictmap_cache.resize(nc);

  /* Phase 1 — NO table lock: decompress, parse, validate, materialize and
   * upload are all batch-local (the arena has its own mutex), so concurrent
   * puts from many ingest threads overlap their heavy work.  Only the
   * global-dictionary interning, delta decode (which interns) and the
   * batches-vector append need t->mu (phase 2 below). */
  for (int c = 0; c < nc; c++) {
    int32_t rc_pc = process_encoded_col(e, t, b, c, columns[c]);
    if (rc_pc != SN_OK) return rc_pc;
  }


  { int32_t rc_ = apply_delete_mask(e, b, delete_mask); if (rc_ != SN_OK) return rc_; }

  apply_stats(t, b, stats);

  /* Phase 2 — under t->mu: global-dictionary interning, delta decode
   * (interns new entries) and the append */
  std::lock_guard<std::mutex> g(t->mu);
  for (int c = 0; c < nc; c++) {
    if (t->schema[c].dtype != SN_TYPE_STRING || b.cols[c].dict.empty())
      continue;
    auto &l2g = b.cols[c].local2global;
    l2g.reserve(b.cols[c].dict.size());
    for (auto &s : b.cols[c].dict) {
      auto it = t->gdict_idx[c].find(s);
      int32_t gid;
      if (it == t->gdict_idx[c].end()) {
        gid = (int32_t)t->gdict[c].size();
        t->gdict[c].push_back(s);
        t->gdict_idx[c].emplace(s, gid);
        if ((int32_t)s.size() > t->gdict_maxlen[c])
          t->gdict_maxlen[c] = (int32_t)s.size();
      } else gid = it->second;
      l2g.push_back(gid);
    }
  }

  { int32_t rc_ = apply_deltas(e, t, b, deltas); if (rc_ != SN_OK) return rc_; }

  t->total_rows += b.num_rows;
  t->batches.push_back(std::move(b));
  return SN_OK;
}

/* Attach the CURRENT mutation state to an existing batch — the seam the
 * reference exercises when UPDATE/DELETE statements write delta blobs and
 * delete masks to the region entries of a batch that already exists
 * (ColumnDelta.scala:300-301 key addressing; ColumnBatchIterator then hands
 * the latest buffers to every scan).  delete_mask and the delta pairs are
 * CUMULATIVE (the reference merges delta chains upward), so they replace
 * any previous state; stats, when given, replace the stats row (the
 * reference rewrites it with a negative batchCount). */
extern "C" int32_t sn_batch_mutate(sn_engine *e, int32_t table, int64_t uuid,
                                   int32_t bucket_id,
                                   const sn_buf *delete_mask,
                                   const sn_buf *deltas,
                                   const sn_buf *stats) {
  Table *t = get_table(e, table);
  if (!t) return fail(SN_ERR_BADARG, "bad table");
  if (e->cfg.shard_count > 1 &&
      (bucket_id % e->cfg.shard_count) != e->cfg.shard_rank)
    return SN_OK;
  std::lock_guard<std::mutex> g(t->mu);
  Batch *b = nullptr;
  for (auto &bb : t->batches)
    if (bb.uuid == uuid && bb.bucket == bucket_id) { b = &bb; break; }
  if (!b) return fail(SN_ERR_BADARG, "no batch uuid=%lld bucket=%d",
                      (long long)uuid, bucket_id);
  if (delete_mask) {
    int32_t rc = apply_delete_mask(e, *b, delete_mask);
    if (rc != SN_OK) return rc;
  }
  if (deltas) {
    int32_t rc = apply_deltas(e, t, *b, deltas);
    if (rc != SN_OK) return rc;
  }
  if (b->stats_pending) {
    /* caller-provided (or absent) stats supersede the device-computed ones */
    b->stats_pending = false;
    if (b->stats_dev) {
      e->arena.release((void *)b->stats_dev,
                       t->schema.size() * 16);
      b->stats_dev = nullptr;
    }
  }
  if (stats) apply_stats(t, b[0], stats);
  else b->stats_valid = false;   /* old bounds no longer trustworthy */
  /* cached descriptor sets reference the old patch/delete device data */
  for (auto &dc : t->desc_caches) {
    e->arena.release((void *)dc.db_dev, dc.db_bytes);
    e->arena.release((void *)dc.tl_dev, dc.tl_bytes);
  }
  t->desc_caches.clear();
  return SN_OK;
}

struct TableInfoProbe { int32_t ncols; sn_type_t dtypes[64]; uint8_t nullable[64]; };
extern "C" int32_t sn_table_schema(sn_engine *e, int32_t table, TableInfoProbe *out) {
  Table *t = get_table(e, table);
  if (!t || !out) return SN_ERR_BADARG;
  out->ncols = (int32_t)t->schema.size();
  for (size_t i = 0; i < t->schema.size(); i++) {
    out->dtypes[i] = t->schema[i].dtype;
    out->nullable[i] = (uint8_t)t->schema[i].nullable;
  }
  return SN_OK;
}
extern "C" void sn_engine_shard(sn_engine *e, int32_t *rank, int32_t *count) {
  *rank = e ? e->cfg.shard_rank : 0;
  *count = e ? e->cfg.shard_count : 1;
}

extern "C" int64_t sn_table_num_batches(sn_engine *e, int32_t table) {
  Table *t = get_table(e, table);
  return t ? (int64_t)t->batches.size() : -1;
}
extern "C" int64_t sn_table_num_rows(sn_engine *e, int32_t table) {
  Table *t = get_table(e, table);
  return t ? t->total_rows : -1;
}

/* fetch back an engine-stored blob (tests / cross-validation) */
extern "C" int64_t sn_table_get_blob(sn_engine *e, int32_t table, int32_t batch,
                                     int32_t col, void *out, int64_t cap) {
  Table *t = get_table(e, table);
  if (!t || batch < 0 || batch >= (int32_t)t->batches.size()) return SN_ERR_BADARG;
  Batch &b = t->batches[batch];
  if (col < 0 || col >= (int32_t)b.cols.size()) return SN_ERR_BADARG;
  if (e->arena.device < 0) {
    auto &hb = b.host_blobs[col];
    if ((int64_t)hb.size() > cap) return SN_ERR_NOMEM;
    memcpy(out, hb.data(), hb.size());
    return (int64_t)hb.size();
  }
  return SN_ERR_UNSUPPORTED;
}

/* ================================================================== */
/* query plane                                                         */

struct GroupOut {
  std::string keys[SN_MAX_GROUPS];
  bool key_null[SN_MAX_GROUPS] = { false, false };
  double sums[SN_MAX_AGGS] = { 0 };
  double counts[SN_MAX_AGGS] = { 0 };
  double rowcount = 0;
};

struct sn_query {
  sn_engine *e = nullptr;
  Table *t = nullptr;
  sn_plan plan;
  int cslot_of_col[64];                 /* table col -> device col slot */
  std::vector<int32_t> used_cols;       /* cslot -> table col */
  int nslots = 0;
  int g1cap = 0, g2cap = 0;             /* per-group-col slot counts */
  int fact_slots = 1;                   /* composite dim_attr x fact_col:
                                           dense fact span (slot minor) */
  bool sparse = false;                  /* open-address hash-aggregate mode */
  bool pac = false;                     /* per-agg counts (nullable agg inputs) */
  bool mm = false;                      /* plan has MIN/MAX aggregates */
  /* compacted group keys + [n][na1] accumulator rows.  Raw arrays, NOT
   * vectors: resize() would zero-fill the ~24 MB 1M-group readback before
   * the copy overwrites it (measured ~1 ms/query of pure host overhead) */
  std::unique_ptr<long long[]> sparse_keys;
  std::unique_ptr<double[]> sparse_rows;
  size_t sparse_n = 0;
  std::vector<double> sparse_null_row;  /* NULL-key group accumulator */
  bool gint[2] = { false, false };      /* integer group key (stats-ranged) */
  int64_t gmin[2] = { 0, 0 };           /* integer key minimum (slot base) */
  int gnull1 = -1, gnull2 = -1;         /* null slot index per group col (-1: none) */
  bool grouped_nonnull_ok = true;
  /* grouped-mode device aggregate dedup: logical agg -> device sweep index
   * (-1 = COUNT(*), derived from the per-slot rowcount) */
  int agg_map[SN_MAX_AGGS];
  int dev_naggs = 0;
  Dim *join_dim = nullptr;          /* group-by-dim-attr key source */
  bool join_group = false;
  /* device buffers */
  double *dev_out = nullptr;
  size_t out_stride = 0;                /* 2*NA_t+1 of the launched template */
  int na_t = 0;                          /* template NAGGS actually launched */
  std::vector<double> host_out;
  int64_t rows_scanned = 0;             /* host metric: rows in unskipped batches */
  int64_t batches_seen = 0, batches_skipped = 0;
  hipEvent_t ev_start = nullptr, ev_stop = nullptr;  /* brackets the scan kernel */
  /* per-query arena blocks, recycled via the free list at destroy */
  std::vector<std::pair<void *, size_t>> owned;
  float kernel_ms = -1.0f;
  bool used_jit = false;                /* query-compiled kernel ran */
  bool done = false;
  bool merged = false;
  std::vector<GroupOut> final_groups;
  int status = SN_OK;
  ~sn_query() {
    if (e) { e->ev_release(ev_start); e->ev_release(ev_stop); ev_start = ev_stop = nullptr; }
    if (ev_start) (void)hipEventDestroy(ev_start);
    if (ev_stop) (void)hipEventDestroy(ev_stop);
  }
};

/* grouped partial-block slot layout (see include/snappy_engine.h) */
struct PartialSlot {
  char keys[SN_MAX_GROUPS][SN_KEY_MAX];
  uint8_t key_null[SN_MAX_GROUPS];
  uint8_t pad[6];
  double sums[SN_MAX_AGGS];
  double counts[SN_MAX_AGGS];
  double rowcount;
};
static_assert(sizeof(PartialSlot) == SN_MAX_GROUPS * SN_KEY_MAX + 8 +
              2 * SN_MAX_AGGS * 8 + 8, "partial slot layout");

static int template_naggs(int na) { return na <= 2 ? 2 : na <= 4 ? 4 : na <= 8 ? 8 : 12; }

static bool batch_skippable(const Batch &b, const sn_plan *p, const Table *t) {
  if (!b.stats_valid || b.has_deltas || b.has_deletes) return false;
  for (int i = 0; i < p->npreds; i++) {
    const sn_pred &pr = p->preds[i];
    int c = pr.col;
    if (b.null_count[c] >= b.num_rows && b.num_rows > 0) return true;
    if (b.bounds_null[c]) continue;
    sn_type_t dt = t->schema[c].dtype;
    bool is_d = dt == SN_TYPE_DOUBLE || dt == SN_TYPE_FLOAT;
    if (is_d) {
      if (pr.has_lo && (pr.lo_strict ? b.hi_d[c] <= pr.lo_d : b.hi_d[c] < pr.lo_d)) return true;
      if (pr.has_hi && (pr.hi_strict ? b.lo_d[c] >= pr.hi_d : b.lo_d[c] > pr.hi_d)) return true;
    } else {
      if (pr.has_lo && (pr.lo_strict ? b.hi_i[c] <= pr.lo_i : b.hi_i[c] < pr.lo_i)) return true;
      if (pr.has_hi && (pr.hi_strict ? b.lo_i[c] >= pr.hi_i : b.lo_i[c] > pr.hi_i)) return true;
    }
  }
  return false;
}

/* engine-cached device workspace for the open-address hash aggregate */
static int ensure_sparse_ws(sn_engine *e, int cap_log2, int naggs1) {
  size_t cap = 1ull << cap_log2;
  if (!e->hws_keys || e->hws_cap_log2 < cap_log2) {
    e->hws_keys = (long long *)e->arena.alloc(cap * 8);
    e->hws_okeys = (long long *)e->arena.alloc((cap + 1) * 8);
    e->hws_cap_log2 = cap_log2;
    e->hws_acc_bytes = 0;
    if (!e->hws_keys || !e->hws_okeys) return SN_ERR_NOMEM;
  }
  size_t accb = ((1ull << e->hws_cap_log2) + 2) * (size_t)naggs1 * 8;
  if (e->hws_acc_bytes < accb) {
    e->hws_acc = (double *)e->arena.alloc(accb);
    e->hws_orows = (double *)e->arena.alloc(accb);
    e->hws_acc_bytes = accb;
    if (!e->hws_acc || !e->hws_orows) return SN_ERR_NOMEM;
  }
  if (!e->hws_flags) {
    e->hws_flags = (int32_t *)e->arena.alloc(64);
    if (!e->hws_flags) return SN_ERR_NOMEM;
  }
  return SN_OK;
}

extern "C" sn_query *sn_query_submit(sn_engine *e, const sn_plan *plan) {
  if (!e || !plan) { fail(SN_ERR_BADARG, "null engine/plan"); return nullptr; }
  Table *t = get_table(e, plan->table);
  if (!t) { fail(SN_ERR_BADARG, "unknown table %d", plan->table); return nullptr; }
  if (!e->has_gpu) { fail(SN_ERR_NOGPU, "no HIP device — the engine never falls back to CPU"); return nullptr; }
  /* one submit at a time per engine (shared stream/scratch/workspaces) */
  std::lock_guard<std::mutex> qg(e->query_mu);
  if (plan->npreds > SN_MAX_PREDS || plan->naggs > SN_MAX_AGGS ||
      plan->ngroup > SN_MAX_GROUPS || plan->naggs <= 0) {
    fail(SN_ERR_BADARG, "plan limits exceeded"); return nullptr;
  }

  auto q = std::make_unique<sn_query>();
  q->e = e; q->t = t; q->plan = *plan;
  for (int i = 0; i < 64; i++) q->cslot_of_col[i] = -1;

  /* collect referenced columns -> device col slots */
  auto use_col = [&](int c) -> int {
    if (c < 0 || c >= (int)t->schema.size()) return -1;
    if (q->cslot_of_col[c] < 0) {
      if ((int)q->used_cols.size() >= SN_DEV_MAX_COLS) return -1;
      q->cslot_of_col[c] = (int)q->used_cols.size();
      q->used_cols.push_back(c);
    }
    return q->cslot_of_col[c];
  };
  for (int i = 0; i < plan->npreds; i++) {
    const sn_pred &pr = plan->preds[i];
    if (use_col(pr.col) < 0) { fail(SN_ERR_BADARG, "bad pred col"); return nullptr; }
    if (t->schema[pr.col].dtype == SN_TYPE_STRING &&
        (!pr.str_eq || pr.str_len <= 0) && pr.in_n <= 0) {
      fail(SN_ERR_UNSUPPORTED,
           "string predicate supports dictionary equality (str_eq) and "
           "IN-lists (in_s) only");
      return nullptr;
    }
  }
  for (int i = 0; i < plan->ngroup; i++) {
    int c = plan->group_cols[i];
    sn_type_t gdt = t->schema[c].dtype;
    bool int_key = gdt == SN_TYPE_INT32 || gdt == SN_TYPE_INT16 ||
                   gdt == SN_TYPE_INT64;
    if (gdt != SN_TYPE_STRING && !int_key) {
      fail(SN_ERR_UNSUPPORTED,
           "group-by supports dictionary string and int16/int32/int64 key columns");
      return nullptr;
    }
    if (int_key && t->schema[c].nullable && plan->ngroup != 1) {
      /* a NULL in one key of a pair is a distinct composite group the
       * packed sparse key cannot represent */
      fail(SN_ERR_UNSUPPORTED,
           "nullable integer group keys supported for single-column "
           "group-bys only");
      return nullptr;
    }
    if (use_col(c) < 0) { fail(SN_ERR_BADARG, "bad group col"); return nullptr; }
  }
  if (plan->ngroup == 2 && plan->group_cols[0] == plan->group_cols[1]) {
    /* both keys share one device column slot, which cannot carry two
     * different premultiplied dictionary images — found by the parity
     * fuzzer producing off-diagonal keys.  GROUP BY a, a is degenerate;
     * reject loudly so the planner dedups upstream. */
    fail(SN_ERR_UNSUPPORTED, "duplicate group column (dedup upstream)");
    return nullptr;
  }
  for (int a = 0; a < plan->naggs; a++)
    for (int f = 0; f < plan->aggs[a].nfactors; f++) {
      int c = plan->aggs[a].factors[f].col;
      sn_type_t dt = t->schema[c].dtype;
      if (dt == SN_TYPE_STRING) { fail(SN_ERR_UNSUPPORTED, "string agg input"); return nullptr; }
      if (use_col(c) < 0) { fail(SN_ERR_BADARG, "too many plan columns"); return nullptr; }
    }

  /* broadcast-dimension join */
  Dim *jd = nullptr;
  if (plan->join_dim != SN_JOIN_NONE) {
    if (plan->join_dim < 0 || plan->join_dim >= (int32_t)e->dims.size()) {
      fail(SN_ERR_BADARG, "unknown dimension %d", plan->join_dim); return nullptr;
    }
    jd = e->dims[plan->join_dim].get();
    if (jd->keys.empty() && !jd->dev_keys) {
      fail(SN_ERR_BADARG, "empty dimension"); return nullptr;
    }
    sn_type_t kt = t->schema[plan->join_fact_col].dtype;
    if (kt != SN_TYPE_INT32 && kt != SN_TYPE_INT64) {
      fail(SN_ERR_UNSUPPORTED, "join key must be int32/int64"); return nullptr;
    }
    if (plan->join_mode == SN_JOIN_GROUP && plan->ngroup > 1) {
      fail(SN_ERR_UNSUPPORTED,
           "dim-attr grouping supports at most ONE fact group column "
           "(attr + fact key fill the 2-key result row)");
      return nullptr;
    }
    if (use_col(plan->join_fact_col) < 0) {
      fail(SN_ERR_BADARG, "too many plan columns"); return nullptr;
    }
    if (plan->join_mode == SN_JOIN_GROUP && jd->attr_maxlen > SN_KEY_MAX - 1) {
      fail(SN_ERR_UNSUPPORTED,
           "dimension attribute exceeds the %d-byte group-key limit",
           SN_KEY_MAX - 1);
      return nullptr;
    }
    if (dim_device_table(e, jd) != SN_OK) return nullptr;
    q->join_dim = jd;
    q->join_group = plan->join_mode == SN_JOIN_GROUP;
  }

  /* ONE t->mu critical section from group-slot derivation through the
   * descriptor build below: a concurrent sn_batch_put (config-5
   * ingest+scan) may grow the global dictionary and append batches; slot
   * capacities, premultiplied dictmaps and the descriptor set must all see
   * the same consistent table state or grouped kernels write accumulator
   * rows out of bounds. */
  std::lock_guard<std::mutex> g(t->mu);
  sync_pending_stats(e, t);   /* raw-ingested batches: resolve device bounds */

  /* group slot space: global dict sizes, plus a null slot only when the
   * schema allows null keys (non-nullable key columns waste no slots —
   * keeps Q1's 3x2 keys in the register-friendly 8x8 kernel) */
  /* integer group keys: the slot space is the stats-derived value range
     (the DictionaryOptimizedMapAccessor direct-slot idea applied to ints).
     Requires every batch to carry valid bounds for the key column and no
     update patches on it (patches may move values outside the bounds). */
  /* dense-slot span of an integer key column, or -1 when the dense path is
   * unavailable (missing stats bounds, update patches, overflowing span) —
   * the caller then falls back to the open-address hash aggregate */
  auto int_key_span = [&](int c, int64_t *mn_out) -> int64_t {
    int64_t mn = 0, mx = -1;
    bool first = true;
    for (auto &b : t->batches) {
      if (!b.stats_valid || b.bounds_null[c]) return -1;
      if (b.had_patches.size() > (size_t)c && b.had_patches[c]) return -1;
      if (first) { mn = b.lo_i[c]; mx = b.hi_i[c]; first = false; }
      else {
        mn = b.lo_i[c] < mn ? b.lo_i[c] : mn;
        mx = b.hi_i[c] > mx ? b.hi_i[c] : mx;
      }
    }
    if (first) { *mn_out = 0; return 1; }   /* empty table */
    *mn_out = mn;
    return mx - mn + 1;
  };
  if (plan->ngroup >= 1) {
    /* group keys travel as fixed SN_KEY_MAX-byte strings in results and
     * partial blocks; distinct keys sharing a 47-byte prefix would silently
     * merge — reject loudly instead (the reference has no key-length limit,
     * so this is a declared engine restriction, not silent divergence) */
    for (int i = 0; i < plan->ngroup; i++) {
      int gc = plan->group_cols[i];
      if (t->schema[gc].dtype == SN_TYPE_STRING &&
          t->gdict_maxlen[gc] > SN_KEY_MAX - 1) {
        fail(SN_ERR_UNSUPPORTED,
             "group key col %d has a %d-byte dictionary entry > %d-byte key limit",
             gc, t->gdict_maxlen[gc], SN_KEY_MAX - 1);
        return nullptr;
      }
    }
    /* try the dense direct-slot space (dictionary ids / stats-ranged
     * integers); integer keys whose span is unbounded, statless, patched or
     * beyond SN_BIG_GROUP_CAP fall back to the open-address hash aggregate
     * (the ByteBufferHashMap/SHAMap analogue, k_grouped_hash) */
    bool col_sparse[2] = { false, false };
    bool any_string = false;
    int caps[2] = { 1, 1 };
    for (int i = 0; i < plan->ngroup; i++) {
      int c = plan->group_cols[i];
      sn_type_t gdt = t->schema[c].dtype;
      if (gdt == SN_TYPE_STRING) {
        any_string = true;
        caps[i] = (int)t->gdict[c].size() + (t->schema[c].nullable ? 1 : 0);
        continue;
      }
      int64_t span = (gdt == SN_TYPE_INT64 || t->schema[c].nullable)
                         ? -1 : int_key_span(c, &q->gmin[i]);
      if (span < 0 || span > SN_BIG_GROUP_CAP) { col_sparse[i] = true; continue; }
      q->gint[i] = true;
      caps[i] = (int)span;
    }
    int64_t dense_slots = (int64_t)std::max(caps[0], 1) *
                          (int64_t)std::max(caps[1], 1);
    q->sparse = col_sparse[0] || col_sparse[1] ||
                (!any_string && dense_slots > SN_BIG_GROUP_CAP);
    if (q->sparse) {
      if (any_string) {
        fail(SN_ERR_UNSUPPORTED,
             "mixed string + sparse integer group keys not supported");
        return nullptr;
      }
      if (plan->ngroup == 2 &&
          (t->schema[plan->group_cols[0]].dtype == SN_TYPE_INT64 ||
           t->schema[plan->group_cols[1]].dtype == SN_TYPE_INT64)) {
        fail(SN_ERR_UNSUPPORTED,
             "two-column group keys with an int64 column not supported "
             "(keys pack into one 64-bit word)");
        return nullptr;
      }
      if (jd) {
        fail(SN_ERR_UNSUPPORTED, "join combined with sparse group keys");
        return nullptr;
      }
      q->gint[0] = q->gint[1] = false;
      q->nslots = 0;
      q->g1cap = q->g2cap = 1;
    } else {
      q->g1cap = caps[0];
      q->g2cap = plan->ngroup == 2 ? caps[1] : 1;
      if (!q->gint[0] && t->schema[plan->group_cols[0]].nullable)
        q->gnull1 = q->g1cap - 1;
      if (plan->ngroup == 2 && !q->gint[1] &&
          t->schema[plan->group_cols[1]].nullable)
        q->gnull2 = q->g2cap - 1;
      if (q->g1cap == 0) q->g1cap = 1;
      if (q->g2cap == 0) q->g2cap = 1;
      q->nslots = q->g1cap * q->g2cap;
      if (q->nslots > SN_BIG_GROUP_CAP) {
        fail(SN_ERR_UNSUPPORTED, "group cardinality %d > %d",
             q->nslots, SN_BIG_GROUP_CAP);
        return nullptr;
      }
    }
  } else if (q->join_group) {
    q->nslots = (int)q->join_dim->attr_dict.size();
    if (q->nslots > SN_MAX_GROUP_SLOTS) {
      fail(SN_ERR_UNSUPPORTED, "dim attr cardinality %d > %d",
           q->nslots, SN_MAX_GROUP_SLOTS);
      return nullptr;
    }
  } else {
    q->nslots = 1;
  }
  if (q->join_group && plan->ngroup >= 1) {
    /* composite GROUP BY dim_attr, fact_col: attr-major slots over the
     * dense fact slot space (sparse fact keys were rejected above) */
    const int attr_cap = (int)q->join_dim->attr_dict.size();
    if (attr_cap > SN_MAX_GROUP_SLOTS) {
      fail(SN_ERR_UNSUPPORTED, "dim attr cardinality %d > %d",
           attr_cap, SN_MAX_GROUP_SLOTS);
      return nullptr;
    }
    q->fact_slots = q->nslots;
    const int64_t total = (int64_t)attr_cap * q->fact_slots;
    if (total > SN_BIG_GROUP_CAP) {
      fail(SN_ERR_UNSUPPORTED, "attr x fact group cardinality %lld > %d",
           (long long)total, SN_BIG_GROUP_CAP);
      return nullptr;
    }
    q->nslots = (int)total;
  }
  /* (na_t set after device-aggregate dedup below) */

  /* build device plan — canonical branchless forms:
   * predicates as closed intervals (strictness folded via nextafter / +-1,
   * missing bounds as +-inf), aggregates as three neutral-padded factors */
  sn_dev_plan dp;
  memset(&dp, 0, sizeof(dp));
  dp.naggs = plan->naggs;
  dp.ngroup = plan->ngroup; dp.nslots = q->nslots;
  dp.nused = (int32_t)q->used_cols.size();
  dp.i64_mask = 0;
  for (size_t ui = 0; ui < q->used_cols.size(); ui++)
    if (t->schema[q->used_cols[ui]].dtype == SN_TYPE_INT64)
      dp.i64_mask |= 1u << ui;
  for (int i = 0; i < plan->ngroup; i++) dp.gcol[i] = q->cslot_of_col[plan->group_cols[i]];
  dp.gmul0 = 1;
  if (plan->ngroup >= 1 && q->gint[0]) {
    dp.gbase[0] = q->gmin[0];
    dp.gmul0 = q->g2cap;       /* dict col0 premultiplies via its dictmap */
  }
  if (plan->ngroup == 2 && q->gint[1]) dp.gbase[1] = q->gmin[1];
  if (jd) {
    dp.jkeys = jd->dev_keys;
    dp.jpayload = jd->dev_payload;
    dp.jcap_log2 = jd->cap_log2;
    dp.jcslot = q->cslot_of_col[plan->join_fact_col];
    dp.jmode = plan->join_mode == SN_JOIN_GROUP ? 1 : 0;
    if (q->join_group && plan->ngroup >= 1)
      dp.jslot_mul = q->fact_slots;   /* composite dim_attr x fact_col */
    dp.jlut = jd->dev_lut;
    dp.jlut_min = jd->lut_min;
    dp.jlut_max = jd->lut_max;
  }
  for (int i = 0; i < plan->npreds; i++) {
    const sn_pred &s = plan->preds[i];
    sn_type_t dt = t->schema[s.col].dtype;
    int cslot = q->cslot_of_col[s.col];
    if (s.in_n > 0) {
      /* IN-list membership (Q12/Q19-class): integer values directly,
       * dictionary strings resolved to premultiplied global ids (absent
       * literals drop out).  Dense value spans build a bitmap LUT; wide
       * spans keep a sorted list (binary-searched; interpreted kernels). */
      if (dp.npreds_in >= 2) { fail(SN_ERR_BADARG, "too many IN predicates"); return nullptr; }
      if (s.has_lo || s.has_hi) {
        fail(SN_ERR_BADARG, "an IN predicate may not also carry range bounds");
        return nullptr;
      }
      std::vector<int64_t> vals;
      if (dt == SN_TYPE_STRING) {
        if (!s.in_s || !s.in_s_len) { fail(SN_ERR_BADARG, "IN needs in_s"); return nullptr; }
        bool is_g2 = plan->ngroup == 2 && plan->group_cols[1] == s.col;
        int64_t mul = is_g2 ? 1 : (q->g2cap > 0 ? q->g2cap : 1);
        for (int32_t vi = 0; vi < s.in_n; vi++) {
          auto it = t->gdict_idx[s.col].find(
              std::string(s.in_s[vi], (size_t)s.in_s_len[vi]));
          if (it != t->gdict_idx[s.col].end())
            vals.push_back((int64_t)it->second * mul);
        }
      } else {
        if (!s.in_i) { fail(SN_ERR_BADARG, "IN needs in_i"); return nullptr; }
        vals.assign(s.in_i, s.in_i + s.in_n);
      }
      std::sort(vals.begin(), vals.end());
      vals.erase(std::unique(vals.begin(), vals.end()), vals.end());
      if (vals.empty()) {
        /* no literal matches anything: impossible range */
        sn_dev_pred_d &d = dp.preds_d[dp.npreds_d++];
        d.cslot = cslot; d.lo = 1.0; d.hi = 0.0;
        continue;
      }
      auto &ip = dp.inp[dp.npreds_in++];
      memset(&ip, 0, sizeof(ip));
      ip.cslot = cslot;
      const int64_t mn = vals.front(), mx = vals.back();
      if (mx - mn < (1ll << 20)) {
        const int64_t nwords = ((mx - mn) >> 6) + 1;
        std::vector<uint64_t> bm((size_t)nwords, 0);
        for (int64_t v : vals) bm[(v - mn) >> 6] |= 1ull << ((v - mn) & 63);
        void *bmd = e->arena.alloc(bm.size() * 8);
        if (!bmd || hipMemcpy(bmd, bm.data(), bm.size() * 8,
                              hipMemcpyHostToDevice) != hipSuccess) {
          fail(SN_ERR_NOMEM, "IN bitmap upload"); return nullptr;
        }
        q->owned.push_back({ bmd, bm.size() * 8 });
        ip.bm = (const uint64_t *)bmd;
        ip.base = mn;
        ip.nwords = (int32_t)nwords;
      } else {
        void *ld = e->arena.alloc(vals.size() * 8);
        if (!ld || hipMemcpy(ld, vals.data(), vals.size() * 8,
                             hipMemcpyHostToDevice) != hipSuccess) {
          fail(SN_ERR_NOMEM, "IN list upload"); return nullptr;
        }
        q->owned.push_back({ ld, vals.size() * 8 });
        ip.list = (const int64_t *)ld;
        ip.n = (int32_t)vals.size();
      }
      continue;
    }
    if (dt == SN_TYPE_STRING) {
      /* dictionary pushdown: literal -> global dict id, compared against
       * the (possibly premultiplied) id the conversion pass writes */
      bool is_g2 = plan->ngroup == 2 && plan->group_cols[1] == s.col;
      int64_t mul = is_g2 ? 1 : (q->g2cap > 0 ? q->g2cap : 1);
      int64_t gid = -1;
      {
        auto it = t->gdict_idx[s.col].find(
            std::string(s.str_eq, (size_t)s.str_len));
        if (it != t->gdict_idx[s.col].end()) gid = it->second;
      }
      sn_dev_pred_d &d = dp.preds_d[dp.npreds_d++];
      d.cslot = cslot;
      if (gid < 0) { d.lo = 1.0; d.hi = 0.0; }   /* literal absent: no rows */
      else { d.lo = d.hi = (double)(gid * mul); }
      continue;
    }
    if (dt == SN_TYPE_INT64) {
      if (dp.npreds_i >= 4) return (fail(SN_ERR_BADARG, "too many int64 preds"), nullptr);
      sn_dev_pred_i &d = dp.preds_i[dp.npreds_i++];
      d.cslot = cslot;
      d.lo = s.has_lo ? (s.lo_strict && s.lo_i < INT64_MAX ? s.lo_i + 1 : s.lo_i)
                      : INT64_MIN;
      d.hi = s.has_hi ? (s.hi_strict && s.hi_i > INT64_MIN ? s.hi_i - 1 : s.hi_i)
                      : INT64_MAX;
    } else {
      bool is_d = (dt == SN_TYPE_DOUBLE || dt == SN_TYPE_FLOAT);
      sn_dev_pred_d &d = dp.preds_d[dp.npreds_d++];
      d.cslot = cslot;
      double lo = s.has_lo ? (is_d ? s.lo_d : (double)s.lo_i) : -INFINITY;
      double hi = s.has_hi ? (is_d ? s.hi_d : (double)s.hi_i) : INFINITY;
      if (s.has_lo && s.lo_strict) lo = std::nextafter(lo, INFINITY);
      if (s.has_hi && s.hi_strict) hi = std::nextafter(hi, -INFINITY);
      d.lo = lo; d.hi = hi;
    }
  }
  {
    /* canonicalize each aggregate; in grouped mode dedupe identical
     * expressions and fold COUNT(*) into the per-slot rowcount, so e.g.
     * Q1's 8 logical aggregates run 5 device sweeps (sum/avg pairs share) */
    const bool grouped = plan->ngroup > 0 || q->join_group;
    int ndev = 0;
    for (int a = 0; a < plan->naggs; a++) {
      sn_dev_agg da;
      da.a0 = da.a1 = da.a2 = 1.0;
      da.m0 = da.m1 = da.m2 = 0.0;
      da.c0 = da.c1 = da.c2 = 0;
      da.nf = 0;
      da._p2 = 0;
      const sn_agg &sa = plan->aggs[a];
      da.op = sa.kind == SN_AGG_MIN ? 1 : sa.kind == SN_AGG_MAX ? 2 : 0;
      if (da.op != 0 && sa.nfactors < 1) {
        fail(SN_ERR_BADARG, "MIN/MAX needs at least one factor");
        return nullptr;
      }
      if (sa.kind != SN_AGG_COUNT_STAR) {
        /* neutral factors get c = c0 below so their LDS reads CSE away */
        da.nf = sa.nfactors;
        if (sa.nfactors >= 1) {
          da.c0 = q->cslot_of_col[sa.factors[0].col];
          da.a0 = sa.factors[0].add; da.m0 = sa.factors[0].mul;
          /* INT64 factors: kernels convert the raw-bit i64 LDS image to
           * double per i64_mask.  A pure SUM(bigint_col) must stay
           * BIT-EXACT (north star): f64 accumulation of integers is exact
           * while every partial sum < 2^53, which the stats bounds prove;
           * when they cannot, fail loudly rather than round silently.
           * (Affine/multi-factor expressions are double-typed in SQL —
           * the 1e-6 double budget applies, no gate.) */
          if ((dp.i64_mask & (1u << da.c0)) && sa.kind == SN_AGG_SUM &&
              sa.nfactors == 1 && sa.factors[0].add == 0.0 &&
              sa.factors[0].mul == 1.0) {
            int gc = sa.factors[0].col;
            double bound = 0.0;
            bool provable = true;
            for (auto &bb : t->batches) {
              if (!bb.stats_valid || bb.bounds_null[gc] ||
                  (bb.had_patches.size() > (size_t)gc && bb.had_patches[gc])) {
                provable = false;
                break;
              }
              double mx = std::max(std::fabs((double)bb.lo_i[gc]),
                                   std::fabs((double)bb.hi_i[gc]));
              bound += (double)bb.num_rows * mx;
            }
            if (!provable || bound >= 9007199254740992.0 /* 2^53 */) {
              fail(SN_ERR_UNSUPPORTED,
                   "SUM over int64 col %d cannot be proven f64-exact "
                   "(needs stats bounds with rows*max|v| < 2^53)", gc);
              return nullptr;
            }
          }
        }
        if (sa.nfactors >= 2) {
          da.c1 = q->cslot_of_col[sa.factors[1].col];
          da.a1 = sa.factors[1].add; da.m1 = sa.factors[1].mul;
        }
        if (sa.nfactors >= 3) {
          da.c2 = q->cslot_of_col[sa.factors[2].col];
          da.a2 = sa.factors[2].add; da.m2 = sa.factors[2].mul;
        }
        if (sa.nfactors < 2) da.c1 = da.c0;
        if (sa.nfactors < 3) da.c2 = sa.nfactors >= 2 ? da.c1 : da.c0;
      }
      if (grouped && sa.kind == SN_AGG_COUNT_STAR) {
        q->agg_map[a] = -1;
        continue;
      }
      int idx = -1;
      if (grouped) {
        for (int j = 0; j < ndev; j++)
          if (memcmp(&dp.aggs[j], &da, sizeof(da)) == 0) { idx = j; break; }
      }
      if (idx < 0) { idx = ndev; dp.aggs[ndev++] = da; }
      q->agg_map[a] = idx;
    }
    q->dev_naggs = grouped ? ndev : plan->naggs;
    dp.naggs = q->dev_naggs;
  }
  for (int a = 0; a < q->dev_naggs; a++)
    if (dp.aggs[a].op != 0) q->mm = true;

  q->na_t = template_naggs(q->dev_naggs > 0 ? q->dev_naggs : 1);
  q->out_stride = 2 * (size_t)q->na_t + 1;

  /* batch descriptors + tile map.  Expensive at many batches (dict-map
   * premultiply + uploads), so cache per plan signature; a cached set is
   * only reusable when stats-skip removes nothing for THIS plan (skipping
   * is an optimization — correctness is unaffected either way).
   * (t->mu is still held — same critical section as the slot derivation.) */
  uint64_t sig = 1469598103934665603ull;
  auto mix = [&](uint64_t v) { sig ^= v; sig *= 1099511628211ull; };
  for (int32_t c : q->used_cols) mix((uint64_t)c + 1);
  mix((uint64_t)plan->ngroup << 8);
  for (int i = 0; i < plan->ngroup; i++) mix((uint64_t)plan->group_cols[i] + 17);
  mix((uint64_t)(q->g1cap * 131 + q->g2cap));
  mix((uint64_t)(plan->join_dim + 3) * 29 + (uint64_t)plan->join_mode);
  mix((uint64_t)t->batches.size());
  for (auto &c : t->gdict) mix(c.size() * 7919);

  int64_t skip_count = 0;
  for (auto &b : t->batches)
    if (batch_skippable(b, plan, t)) skip_count++;
  q->batches_seen = (int64_t)t->batches.size();
  q->batches_skipped = skip_count;

  DescCache *hit = nullptr;
  if (skip_count == 0) {
    for (auto &dc : t->desc_caches)
      if (dc.sig == sig && dc.batch_count == t->batches.size()) { hit = &dc; break; }
  }

  std::vector<sn_dev_batch> hbatches;
  std::vector<sn_dev_tile> htiles;
  /* JIT eligibility: every unskipped batch clean, kinds uniform + stageable
   * (derived here; stored in the DescCache for reuse on hits) */
  bool jit_ok_b = true, jit_first = true, jit_any_del = false;
  int jit_kinds[SN_DEV_MAX_COLS] = {0};
  /* per-agg counts needed?  Only when a grouped plan's aggregate-input
   * columns carry ACTUAL nulls (null-free nullable schemas keep the
   * counts==rowcount fast layout, JIT included) */
  uint32_t agg_cmask = 0;
  for (int a = 0; a < q->dev_naggs; a++) {
    const sn_dev_agg &A = dp.aggs[a];
    if (A.nf >= 1) agg_cmask |= 1u << A.c0;
    if (A.nf >= 2) agg_cmask |= 1u << A.c1;
    if (A.nf >= 3) agg_cmask |= 1u << A.c2;
  }
  bool agg_nulls = false;
  for (auto &b : t->batches) {
    if (hit) break;
    if (skip_count && batch_skippable(b, plan, t)) continue;
    sn_dev_batch db;
    memset(&db, 0, sizeof(db));
    db.num_rows = b.num_rows;
    db.del_bm = b.del_bm_dev;
    bool clean = !b.has_deletes;
    for (size_t ui = 0; ui < q->used_cols.size(); ui++) {
      int c = q->used_cols[ui];
      const ColMeta &m = b.cols[c];
      sn_dev_col &dc = db.cols[ui];
      memset(&dc, 0, sizeof(dc));
      sn_type_t dt = t->schema[c].dtype;
      dc.body = (const uint8_t *)b.col_dev[c] + m.body_off;
      dc.has_nulls = m.num_null_words > 0;
      if (dc.has_nulls) {
        dc.nullw = (const uint64_t *)((const uint8_t *)b.col_dev[c] + m.null_off);
        dc.nullpfx = b.nullpfx_dev[c];
        clean = false;
      }
      switch (m.type_id) {
        case SN_ENC_UNCOMPRESSED:
          switch (dt) {
            case SN_TYPE_DOUBLE: dc.kind = SN_K_F64; break;
            case SN_TYPE_INT32: dc.kind = SN_K_I32; break;
            case SN_TYPE_INT64: dc.kind = SN_K_I64; break;
            case SN_TYPE_FLOAT: dc.kind = SN_K_F32; break;
            case SN_TYPE_INT16: dc.kind = SN_K_I16; break;
            case SN_TYPE_BOOL: dc.kind = SN_K_U8; break;
            case SN_TYPE_INT8: dc.kind = SN_K_S8; break;
            default:
              fail(SN_ERR_UNSUPPORTED, "uncompressed %d on GPU path", (int)dt);
              return nullptr;
          }
          break;
        case SN_ENC_RUNLENGTH:
          if (b.rle_n[c] <= 0 && b.num_rows > 0) {
            fail(SN_ERR_UNSUPPORTED, "RLE col %d has no run aux", c);
            return nullptr;
          }
          /* INT64 runs carry raw bits in rle_vals (put-time extraction) —
           * a distinct kind so read_general bitcasts instead of rounding */
          dc.kind = dt == SN_TYPE_INT64 ? SN_K_RLE_I64 : SN_K_RLE;
          dc.rle_ends = b.rle_ends_dev[c];
          dc.rle_vals = b.rle_vals_dev[c];
          dc.rle_n = b.rle_n[c];
          break;
        case SN_ENC_BOOLEAN_BITSET:
          dc.kind = SN_K_BOOLBIT;
          break;
        case SN_ENC_DICTIONARY:
        case SN_ENC_BIG_DICTIONARY: {
          if (dt != SN_TYPE_STRING) {
            fail(SN_ERR_UNSUPPORTED, "int dictionary col on GPU path not yet supported");
            return nullptr;
          }
          dc.kind = m.type_id == SN_ENC_DICTIONARY ? SN_K_DICT16 : SN_K_DICT32;
          /* local->premultiplied-global map, cached per (mul, null_gid) —
           * local2global is fixed per batch, so queries with the same
           * group geometry reuse one device map */
          bool is_g2 = plan->ngroup == 2 && plan->group_cols[1] == c;
          int mul = is_g2 ? 1 : (q->g2cap > 0 ? q->g2cap : 1);
          int gn = is_g2 ? q->gnull2 : q->gnull1;
          int null_gid_local = gn >= 0 ? (is_g2 ? gn : gn * q->g2cap) : 0;
          dc.null_gid = null_gid_local;
          auto &mcache = b.dictmap_cache[c];
          auto mit = mcache.find({ mul, null_gid_local });
          if (mit != mcache.end()) {
            dc.dictmap = mit->second;
          } else {
            std::vector<int32_t> map(m.local2global.size() + 1);
            for (size_t i = 0; i < m.local2global.size(); i++)
              map[i] = m.local2global[i] * mul;
            /* index == numElements denotes null (DictionaryEncoding.scala:90) */
            map[m.local2global.size()] = null_gid_local;
            dc.dictmap = (const int32_t *)up(e, map.data(), map.size() * 4);
            mcache.emplace(std::make_pair(mul, null_gid_local), dc.dictmap);
          }
          break;
        }
        default:
          fail(SN_ERR_UNSUPPORTED, "encoding %d on GPU path not yet supported",
               m.type_id);
          return nullptr;
      }
      /* patches */
      if (b.patch_dev[c].n > 0) {
        clean = false;
        const Batch::PatchDev &pd = b.patch_dev[c];
        dc.patch_bm = pd.bm;
        dc.patch_pos = pd.pos;
        dc.patch_nullbm = pd.nullbm;
        dc.patch_n = pd.n;
        if (dt == SN_TYPE_STRING) {
          /* premultiply global ids for this query */
          bool is_g2 = plan->ngroup == 2 && plan->group_cols[1] == c;
          int mul = is_g2 ? 1 : (q->g2cap > 0 ? q->g2cap : 1);
          std::vector<double> pv(b.patch_host[c].val.size());
          for (size_t i = 0; i < pv.size(); i++)
            pv[i] = b.patch_host[c].val[i] * mul;
          dc.patch_val = (const double *)up(e, pv.data(), pv.size() * 8);
        } else {
          dc.patch_val = pd.val;
        }
      }
    }
    db.clean = clean ? 1 : 0;
    for (size_t ui = 0; ui < q->used_cols.size(); ui++)
      if (((agg_cmask >> ui) & 1) &&
          (db.cols[ui].has_nulls || db.cols[ui].patch_nullbm))
        agg_nulls = true;
    /* deletes alone don't disqualify the JIT (the generated kernel reads
     * del_bm); nulls or unmaterialized patches on a used column do */
    if (db.del_bm) jit_any_del = true;
    for (size_t ui = 0; ui < q->used_cols.size() && jit_ok_b; ui++) {
      const sn_dev_col &jc = db.cols[ui];
      int k = jc.kind;
      bool stageable = k == SN_K_F64 || k == SN_K_I64 || k == SN_K_I32 ||
                       k == SN_K_F32 || k == SN_K_I16 || k == SN_K_DICT16 ||
                       k == SN_K_DICT32;
      if (!stageable || jc.has_nulls || jc.patch_n > 0) jit_ok_b = false;
      else if (jit_first) jit_kinds[ui] = k;
      else if (jit_kinds[ui] != k) jit_ok_b = false;
    }
    jit_first = false;
    int32_t bi = (int32_t)hbatches.size();
    hbatches.push_back(db);
    q->rows_scanned += b.num_rows;
    for (int32_t r = 0; r < b.num_rows; r += SN_TILE_ROWS)
      htiles.push_back({ bi, r });
  }

  /* upload + launch */
  size_t out_n = (size_t)q->nslots * q->out_stride;
  if (!q->sparse) {
    q->dev_out = (double *)e->arena.alloc(out_n * 8);
    if (!q->dev_out) { fail(SN_ERR_NOMEM, "out alloc"); return nullptr; }
    q->owned.push_back({ q->dev_out, out_n * 8 });
    if (hipMemsetAsync(q->dev_out, 0, out_n * 8, e->stream) != hipSuccess) {
      fail(SN_ERR_GENERIC, "memset out"); return nullptr;
    }
  }
  const void *db_dev = nullptr, *tl_dev = nullptr;
  int32_t ntiles = 0;
  if (hit) {
    db_dev = hit->db_dev; tl_dev = hit->tl_dev;
    ntiles = hit->ntiles;
    q->rows_scanned = hit->rows;
  } else if (!htiles.empty()) {
    void *dbp = e->arena.alloc(hbatches.size() * sizeof(sn_dev_batch));
    void *tlp = e->arena.alloc(htiles.size() * sizeof(sn_dev_tile));
    if (!dbp || !tlp) { fail(SN_ERR_NOMEM, "desc alloc"); return nullptr; }
    /* blocking copies: the host vectors are stack-local and die at return */
    if (hipMemcpy(dbp, hbatches.data(), hbatches.size() * sizeof(sn_dev_batch),
                  hipMemcpyHostToDevice) != hipSuccess ||
        hipMemcpy(tlp, htiles.data(), htiles.size() * sizeof(sn_dev_tile),
                  hipMemcpyHostToDevice) != hipSuccess) {
      fail(SN_ERR_GENERIC, "desc upload"); return nullptr;
    }
    db_dev = dbp; tl_dev = tlp;
    ntiles = (int32_t)htiles.size();
    if (skip_count != 0) {
      /* not cacheable (stats-skip removed batches): this query owns the
       * descriptor blocks; they recycle at sn_query_destroy */
      q->owned.push_back({ dbp, hbatches.size() * sizeof(sn_dev_batch) });
      q->owned.push_back({ tlp, htiles.size() * sizeof(sn_dev_tile) });
    }
    if (skip_count == 0) {
      if (t->desc_caches.size() > 16) {
        /* evict the oldest cached set (FIFO); stream ordering guarantees
         * any in-flight kernel reading it finishes before a reuse write */
        DescCache &old = t->desc_caches.front();
        e->arena.release((void *)old.db_dev, old.db_bytes);
        e->arena.release((void *)old.tl_dev, old.tl_bytes);
        t->desc_caches.erase(t->desc_caches.begin());
      }
      DescCache dc;
      dc.sig = sig; dc.batch_count = t->batches.size();
      dc.db_dev = db_dev; dc.tl_dev = tl_dev; dc.ntiles = ntiles;
      dc.db_bytes = hbatches.size() * sizeof(sn_dev_batch);
      dc.tl_bytes = htiles.size() * sizeof(sn_dev_tile);
      dc.rows = q->rows_scanned;
      dc.jit_ok = (jit_ok_b && !jit_first) ? 1 : 0;
      dc.jit_del = jit_any_del ? 1 : 0;
      memcpy(dc.jit_kinds, jit_kinds, sizeof(jit_kinds));
      dc.pac = agg_nulls ? 1 : 0;
      t->desc_caches.push_back(dc);
    }
  }
  {
    /* MIN/MAX plans always take the pac layout (per-agg counts carry the
     * empty-group NULL semantics; routing goes through the op-aware
     * grouped kernels, keyless included) */
    const bool grouped_mode = plan->ngroup > 0 || q->join_group;
    q->pac = (grouped_mode && (hit ? hit->pac != 0 : agg_nulls)) || q->mm;
    dp.pac = q->pac ? 1 : 0;
  }
  if (q->sparse && ntiles > 0) {
    /* ---- open-address hash-aggregate launch (k_grouped_hash) ----
     * The workspace (key table + accumulators) is engine-cached; results
     * are compacted on device and read back HERE (inside submit, stream
     * synchronized), so concurrent queries never share the live table. */
    const int naggs1 = (q->pac ? 2 : 1) * q->dev_naggs + 1;
    /* capacity: 2x headroom over the worst-case distinct count keeps the
     * linear probe short (load factor <= 0.5 at full distinctness); the
     * fill counter grows the table before probing turns pathological,
     * and the per-table hint skips the rediscovery next query */
    int cap_log2 = 21;
    while (cap_log2 > 12 && (1ll << (cap_log2 - 1)) >= 2 * q->rows_scanned)
      cap_log2--;                      /* small tables: smaller table */
    if (t->sparse_cap_hint > cap_log2) cap_log2 = t->sparse_cap_hint;
    q->ev_start = e->ev_acquire();
    q->ev_stop = e->ev_acquire();
    /* radix two-pass: the single-pass probe touches ~3 random
     * 64 B lines per row across a table far bigger than L2 (measured
     * 182 B/row of HBM traffic on 1M keys).  For big tables the compiled
     * pass 1 scatters (key, values) records into cap/4096 hash partitions
     * (streaming traffic) and k_radix_agg aggregates each partition into
     * its private L2-resident table segment.  JIT-only (clean batches);
     * pac excluded (records carry values, not per-agg validity). */
    static const bool radix_env = [] {
      const char *v = getenv("SN_RADIX");
      return !v || v[0] != '0';
    }();
    bool radix_ok = radix_env && !q->pac && q->dev_naggs <= 4;
    bool done_h = false;
    for (int attempt = 0; attempt < 4 && !done_h; attempt++) {
      if (ensure_sparse_ws(e, cap_log2, naggs1) != SN_OK) {
        fail(SN_ERR_NOMEM, "sparse hash workspace"); return nullptr;
      }
      const size_t cap = 1ull << cap_log2;
      sn_dev_plan dps = dp;
      dps.sparse = 1;
      dps.hkeys = e->hws_keys;
      dps.hacc = e->hws_acc;
      dps.hflags = e->hws_flags;
      dps.hcap_log2 = cap_log2;
      bool radix = radix_ok && cap_log2 >= 17;
      static const int sub_env = [] {      /* A/B lever for the sweep */
        const char *v = getenv("SN_RADIX_SUB");
        return v ? atoi(v) : 0;
      }();
      const int sub_log2 =
          std::max(sub_env ? sub_env : SN_RADIX_SUB_MIN,
                   cap_log2 - SN_RADIX_NPART_MAX_LOG2);
      if (radix) {
        const int npart = 1 << (cap_log2 - sub_log2);
        const long long percap =
            2 * q->rows_scanned / npart + 4096;
        const size_t recb = (size_t)npart * (size_t)percap *
                            (size_t)(1 + q->dev_naggs) * 8;
        if (percap > INT32_MAX / 2 || recb > (48ull << 30)) {
          radix = false;
        } else {
          if (!e->rws_recs || e->rws_bytes < recb) {
            if (e->rws_recs) e->arena.release(e->rws_recs, e->rws_bytes);
            e->rws_recs = (double *)e->arena.alloc(recb);
            e->rws_bytes = e->rws_recs ? recb : 0;
          }
          if (e->rws_npart < npart) {
            if (e->rws_pcount)
              e->arena.release(e->rws_pcount, (size_t)e->rws_npart * 4);
            e->rws_pcount = (int32_t *)e->arena.alloc((size_t)npart * 4);
            e->rws_npart = e->rws_pcount ? npart : 0;
          }
          if (!e->rws_recs || !e->rws_pcount) {
            radix = false;               /* fall back to the single pass */
          } else {
            dps.radix = sub_log2;
            dps.precs = e->rws_recs;
            dps.pcount = e->rws_pcount;
            dps.percap = (int32_t)percap;
            if (hipMemsetAsync(e->rws_pcount, 0, (size_t)npart * 4,
                               e->stream) != hipSuccess) {
              fail(SN_ERR_GENERIC, "radix counters zero"); return nullptr;
            }
          }
        }
      }
      void *dps_dev = e->arena.alloc(sizeof(dps));
      if (!dps_dev ||
          hipMemcpy(dps_dev, &dps, sizeof(dps), hipMemcpyHostToDevice) != hipSuccess) {
        fail(SN_ERR_NOMEM, "sparse plan upload"); return nullptr;
      }
      q->owned.push_back({ dps_dev, sizeof(dps) });
      if (hipMemsetAsync(e->hws_keys, 0xff, cap * 8, e->stream) != hipSuccess ||
          hipMemsetAsync(e->hws_acc, 0, (cap + 2) * (size_t)naggs1 * 8,
                         e->stream) != hipSuccess ||
          hipMemsetAsync(e->hws_flags, 0, 16, e->stream) != hipSuccess) {
        fail(SN_ERR_GENERIC, "sparse workspace zero"); return nullptr;
      }
      if (q->mm &&
          sn_launch_acc_init(e->hws_acc, (long long)cap + 2, q->dev_naggs,
                             naggs1, dps_dev, e->stream) != 0) {
        fail(SN_ERR_GENERIC, "sparse min/max init"); return nullptr;
      }
      /* query-compiled twin first (plan structure compile-time, capacity
       * tokenized); any miss falls back to the interpreted hash kernel */
      void *jfn = nullptr;
      const bool jit_in_ok_s =
          dp.npreds_in == 0 ||
          (dp.inp[0].bm && (dp.npreds_in < 2 || dp.inp[1].bm));
      if (e->jit && jit_in_ok_s && q->dev_naggs <= 12 &&
          (hit ? hit->jit_ok != 0 : (jit_ok_b && !jit_first))) {
        const int *jk = hit ? hit->jit_kinds : jit_kinds;
        int jdel = hit ? hit->jit_del : (jit_any_del ? 1 : 0);
        jfn = sn_jit_get(e->jit, &dps, jk, 0, q->na_t, jdel);
      }
      if (q->ev_start) (void)hipEventRecord(q->ev_start, e->stream);
      int rc = -1;
      if (jfn) {
        int jgrid;
        if (ntiles <= SN_GRID_CAP) jgrid = ntiles;
        else {
          int rounds = (ntiles + SN_GRID_CAP - 1) / SN_GRID_CAP;
          jgrid = (ntiles + rounds - 1) / rounds;
        }
        rc = sn_jit_launch(jfn, jgrid, (const sn_dev_batch *)db_dev,
                           (const sn_dev_tile *)tl_dev, ntiles,
                           e->hws_acc, (const int64_t *)e->hws_keys,
                           (const int32_t *)e->hws_flags, nullptr,
                           (const sn_dev_plan *)dps_dev, e->stream);
        q->used_jit = rc == 0;
        if (rc != 0) jfn = nullptr;
      }
      if (!jfn)
        rc = sn_launch_hash_scan(&dps, (const sn_dev_plan *)dps_dev,
                                 (const sn_dev_batch *)db_dev,
                                 (const sn_dev_tile *)tl_dev, ntiles,
                                 e->stream);
      /* radix pass 2: aggregate the partitioned records (only meaningful
       * after the compiled pass 1).  The LDS variant compacts straight
       * into okeys/orows via the counter at hws_flags[1]. */
      const bool radix_ran = radix && jfn;
      const bool radix_direct = radix_ran && sn_radix_direct(sub_log2, naggs1);
      if (rc == 0 && radix_ran)
        rc = sn_launch_radix_agg(&dps, (const sn_dev_plan *)dps_dev,
                                 e->hws_okeys, e->hws_orows,
                                 (int *)(e->hws_flags + 1), e->stream);
      if (q->ev_stop) (void)hipEventRecord(q->ev_stop, e->stream);
      if (rc != 0) {
        fail(SN_ERR_GENERIC, "hash-agg launch: %s",
             hipGetErrorString((hipError_t)rc));
        return nullptr;
      }
      if (hipStreamSynchronize(e->stream) != hipSuccess) {
        fail(SN_ERR_GENERIC, "hash-agg sync"); return nullptr;
      }
      int32_t fl[4] = { 0, 0, 0, 0 };
      (void)hipMemcpy(fl, e->hws_flags, 16, hipMemcpyDeviceToHost);
      if (radix_ran && fl[3]) {
        /* record buffer overflow (pathological key skew): redo this
         * attempt through the single-pass probe — nothing was lost, the
         * table is re-zeroed per attempt */
        radix_ok = false;
        attempt--;
        continue;
      }
      const int32_t ovf = fl[0];
      const int64_t fill = fl[2];
      /* grow on hard overflow OR load factor > 0.6 (probe chains degrade
       * sharply past that); results so far are correct either way */
      if (ovf || (fill * 5 > (int64_t)cap * 3 && cap_log2 < 24)) {
        if (ovf && cap_log2 >= 24) {
          fail(SN_ERR_OVERFLOW,
               "sparse group cardinality exceeds the 2^24 hash table");
          return nullptr;
        }
        cap_log2 = std::min(cap_log2 + 2, 24);
        continue;
      }
      t->sparse_cap_hint = std::max(t->sparse_cap_hint, cap_log2);
      /* the LDS radix pass already compacted into okeys/orows (counter at
       * hws_flags[1]); otherwise compact the table here */
      if (!radix_direct &&
          (hipMemsetAsync(e->hws_flags + 1, 0, 4, e->stream) != hipSuccess ||
           sn_launch_hash_compact(e->hws_keys, e->hws_acc, (int)cap, naggs1,
                                  e->hws_okeys, e->hws_orows,
                                  (int *)(e->hws_flags + 1), e->stream) != 0 ||
           hipStreamSynchronize(e->stream) != hipSuccess)) {
        fail(SN_ERR_GENERIC, "hash-agg compact"); return nullptr;
      }
      int32_t ngrp = 0;
      (void)hipMemcpy(&ngrp, e->hws_flags + 1, 4, hipMemcpyDeviceToHost);
      /* the NULL-key group accumulates in the extra row at cap+1 (the
       * compaction covers rows [0, cap]) */
      q->sparse_null_row.assign((size_t)naggs1, 0.0);
      (void)hipMemcpy(q->sparse_null_row.data(),
                      e->hws_acc + (size_t)(cap + 1) * naggs1,
                      (size_t)naggs1 * 8, hipMemcpyDeviceToHost);
      q->sparse_keys.reset(new long long[(size_t)ngrp + 1]);
      q->sparse_rows.reset(new double[((size_t)ngrp + 1) * naggs1]);
      q->sparse_n = (size_t)ngrp;
      if (ngrp > 0) {
        /* plain pageable D2H: measured ~55 GB/s and stable once the heap
         * recycles (mallopt in engine create) — the pinned bounce costs a
         * second 24 MB host memcpy for nothing */
        if (hipMemcpy(q->sparse_keys.get(), e->hws_okeys, (size_t)ngrp * 8,
                      hipMemcpyDeviceToHost) != hipSuccess ||
            hipMemcpy(q->sparse_rows.get(), e->hws_orows,
                      (size_t)ngrp * naggs1 * 8,
                      hipMemcpyDeviceToHost) != hipSuccess) {
          fail(SN_ERR_GENERIC, "hash-agg readback"); return nullptr;
        }
      }
      if (radix_direct) {
        /* the LDS compaction never scans the (untouched) global table, so
         * append the reserved sentinel-key row (a REAL key of -1) here —
         * the same row k_hash_compact emits from index cap (the arrays
         * were sized ngrp+1 for exactly this) */
        double *rrow = q->sparse_rows.get() + q->sparse_n * naggs1;
        (void)hipMemcpy(rrow, e->hws_acc + (size_t)cap * naggs1,
                        (size_t)naggs1 * 8, hipMemcpyDeviceToHost);
        if (rrow[naggs1 - 1] != 0.0) {
          q->sparse_keys[q->sparse_n] = SN_HASH_EMPTY;
          q->sparse_n++;
        }
      }
      done_h = true;
    }
    {
      std::lock_guard<std::mutex> ga(e->aux_mu);
      e->live_q.insert(q.get());
    }
    return q.release();
  }
  if (ntiles > 0) {
    /* HIP events bracket the scan kernel on ITS stream for the roofline leg
     * (torch.cuda.Event would only see torch's current stream) */
    void *dp_dev = nullptr;
    {
      uint64_t dph = 1469598103934665603ull;
      const uint8_t *db_ = (const uint8_t *)&dp;
      for (size_t i = 0; i < sizeof(dp); i++) { dph ^= db_[i]; dph *= 1099511628211ull; }
      std::lock_guard<std::mutex> ga(e->aux_mu);
      auto it = e->dp_cache.find(dph);
      if (it != e->dp_cache.end()) dp_dev = it->second;
      else {
        dp_dev = e->arena.alloc(sizeof(dp));
        if (!dp_dev ||
            hipMemcpy(dp_dev, &dp, sizeof(dp), hipMemcpyHostToDevice) != hipSuccess) {
          fail(SN_ERR_NOMEM, "plan upload"); return nullptr;
        }
        if (e->dp_cache.size() > 64) {
          for (auto &kv : e->dp_cache)
            e->arena.release(kv.second, sizeof(sn_dev_plan));
          e->dp_cache.clear();
        }
        e->dp_cache[dph] = dp_dev;
      }
    }
    /* block-partial scratch rows (the >16-slot path caps its grid) */
    int grid = ntiles < SN_GRID_CAP ? ntiles : SN_GRID_CAP;
    if (dp.nslots > 16 && grid > SN_GRID_BIGSLOT) grid = SN_GRID_BIGSLOT;
    const int na1 = dp.pac ? 2 * q->dev_naggs + 1 : q->dev_naggs + 1;
    size_t nv = (dp.nslots <= 1 && !dp.pac)
                    ? (size_t)(2 * q->na_t + 1)
                    : (size_t)(dp.nslots < 1 ? 1 : dp.nslots) * na1;
    const bool big_groups =
        dp.nslots > SN_RESULT_PAGE ||
        (dp.pac && sn_grouped_needs_global(dp.nused, dp.nslots, na1));
    size_t need = (big_groups ? 8 : (size_t)grid) * nv * 8;
    if (e->scratch_sz < need) {
      e->scratch = (double *)e->arena.alloc(need);
      e->scratch_sz = need;
      if (!e->scratch) { fail(SN_ERR_NOMEM, "scratch alloc"); return nullptr; }
    }
    if (big_groups &&
        hipMemsetAsync(e->scratch, 0, 8 * nv * 8, e->stream) != hipSuccess) {
      fail(SN_ERR_GENERIC, "accumulator zero"); return nullptr;
    }
    if (big_groups && q->mm &&
        sn_launch_acc_init(e->scratch, 8ll * dp.nslots, q->dev_naggs, na1,
                           dp_dev, e->stream) != 0) {
      fail(SN_ERR_GENERIC, "accumulator min/max init"); return nullptr;
    }
    q->ev_start = e->ev_acquire();
    q->ev_stop = e->ev_acquire();
    /* query-compiled kernel (jit.cpp): plan constants baked as literals —
     * the WholeStageCodegen analogue; measured 3x on Q1's shape over the
     * interpreted runtime-plan kernel.  Any miss falls back, still on GPU. */
    void *jfn = nullptr;
    bool jit_shape_ok;
    if (dp.nslots <= 1 && !dp.pac) jit_shape_ok = dp.naggs <= 12;
    else if (dp.nslots <= 8) jit_shape_ok = dp.naggs <= 6;
    else if (dp.nslots <= 1024)
      /* LDS-accumulator mode: static shared = LDS image + gacc must fit */
      jit_shape_ok = (size_t)dp.nused * 8192 +
                     (size_t)dp.nslots * na1 * 8 + 1024 <= 160 * 1024;
    else
      /* global-atomic mode: only the LDS image constrains */
      jit_shape_ok = dp.nslots <= SN_BIG_GROUP_CAP;
    /* pac from ACTUAL nulls never reaches here (jit_ok excludes null
     * batches); pac from MIN/MAX compiles op-aware kernels.  IN-lists run
     * compiled only in bitmap form (sorted-list search stays interpreted). */
    const bool jit_in_ok =
        dp.npreds_in == 0 ||
        (dp.inp[0].bm && (dp.npreds_in < 2 || dp.inp[1].bm));
    if (e->jit && jit_shape_ok && jit_in_ok &&
        (hit ? hit->jit_ok != 0 : (jit_ok_b && !jit_first))) {
      const int *jk = hit ? hit->jit_kinds : jit_kinds;
      int jdel = hit ? hit->jit_del : (jit_any_del ? 1 : 0);
      jfn = sn_jit_get(e->jit, &dp, jk, dp.nslots, q->na_t, jdel);
    }
    if (q->ev_start) (void)hipEventRecord(q->ev_start, e->stream);
    int rc = -1;
    if (jfn) {
      int jgrid;
      if (ntiles <= SN_GRID_CAP) jgrid = ntiles;
      else {
        int rounds = (ntiles + SN_GRID_CAP - 1) / SN_GRID_CAP;
        jgrid = (ntiles + rounds - 1) / rounds;
      }
      /* >16-slot scratch rows are wide; mirror the interpreted grid cap
       * (scratch was sized with the same bound above) */
      if (dp.nslots > 16 && jgrid > SN_GRID_BIGSLOT) jgrid = SN_GRID_BIGSLOT;
      int naggs1 = (dp.nslots <= 1 && !dp.pac) ? 0 : na1;
      rc = sn_jit_launch(jfn, jgrid, (const sn_dev_batch *)db_dev,
                         (const sn_dev_tile *)tl_dev, ntiles, e->scratch,
                         dp.jkeys, dp.jpayload, dp.jlut,
                         (const sn_dev_plan *)dp_dev, e->stream);
      if (rc == 0)
        rc = sn_launch_reduce(e->scratch, big_groups ? 8 : jgrid, (int)nv,
                              q->dev_out, naggs1, (int)q->out_stride,
                              dp_dev, e->stream);
      q->used_jit = rc == 0;
      if (rc != 0) jfn = nullptr;   /* interpreted kernels take over */
    }
    if (!jfn)
      rc = sn_launch_scan_agg(&dp, (const sn_dev_plan *)dp_dev,
                              (const sn_dev_batch *)db_dev,
                              (const sn_dev_tile *)tl_dev, ntiles,
                              q->dev_out, e->scratch, e->stream);
    if (q->ev_stop) (void)hipEventRecord(q->ev_stop, e->stream);
    if (rc != 0) {
      fail(SN_ERR_GENERIC, "kernel launch: %s", hipGetErrorString((hipError_t)rc));
      return nullptr;
    }
  }
  {
    std::lock_guard<std::mutex> ga(e->aux_mu);
    e->live_q.insert(q.get());
  }
  return q.release();
}

extern "C" int32_t sn_query_wait(sn_query *q) {
  if (!q) return SN_ERR_BADARG;
  if (!q->done) {
    HIP_OR_FAIL(hipStreamSynchronize(q->e->stream));
    size_t out_n = (size_t)q->nslots * q->out_stride;
    q->host_out.resize(out_n);
    if (out_n > 0 && q->dev_out)
      HIP_OR_FAIL(hipMemcpy(q->host_out.data(), q->dev_out, out_n * 8,
                            hipMemcpyDeviceToHost));
    if (q->ev_start && q->ev_stop)
      (void)hipEventElapsedTime(&q->kernel_ms, q->ev_start, q->ev_stop);
    q->done = true;
  }
  return SN_OK;
}

/* scan-kernel duration in ms (HIP events on the launch stream); -1 if the
 * query launched nothing */
extern "C" double sn_query_kernel_ms(sn_query *q) {
  if (!q) return -1.0;
  (void)sn_query_wait(q);
  return (double)q->kernel_ms;
}

/* 1 when the query ran a query-compiled (hipRTC) kernel, 0 interpreted */
extern "C" int32_t sn_query_used_jit(sn_query *q) {
  return q && q->used_jit ? 1 : 0;
}

/* number of compiled kernels in this engine's JIT cache (tokenized plan
 * shapes — different literal values share one entry) */
extern "C" int32_t sn_engine_jit_count(sn_engine *e) {
  return e ? sn_jit_cache_count(e->jit) : 0;
}

/* local accumulators -> GroupOut list (pre-merge view) */
static void local_groups(sn_query *q, std::vector<GroupOut> *out) {
  const sn_plan &p = q->plan;
  Table *t = q->t;
  if (q->sparse) {
    /* hash-aggregate results: compacted (key, accumulator-row) pairs.
     * Keys surface as decimal text like the dense integer-key path (the
     * partial-block and result formats stay shared). */
    const int naggs1 = (q->pac ? 2 : 1) * q->dev_naggs + 1;
    const bool packed = p.ngroup == 2;
    auto fill = [&](const double *row) -> GroupOut {
      GroupOut g;
      g.rowcount = row[naggs1 - 1];
      for (int a = 0; a < p.naggs; a++) {
        int di = q->agg_map[a];
        double s = di < 0 ? g.rowcount : row[di];
        int k = p.aggs[a].kind;
        if (di >= 0 && (k == SN_AGG_MIN || k == SN_AGG_MAX)) {
          /* sparse rows bypass k_reduce: decode the ord-u64 cell here */
          unsigned long long o;
          memcpy(&o, &row[di], 8);
          s = sn_ord_f64_h(o);
        }
        g.sums[a] = s;
        g.counts[a] = (di < 0 || !q->pac) ? g.rowcount
                                          : row[q->dev_naggs + di];
      }
      return g;
    };
    for (size_t i = 0; i < q->sparse_n; i++) {
      const double *row = &q->sparse_rows[i * naggs1];
      if (row[naggs1 - 1] == 0.0) continue;
      GroupOut g = fill(row);
      long long key = q->sparse_keys[i];
      if (packed) {
        g.keys[0] = std::to_string((int32_t)((unsigned long long)key >> 32));
        g.keys[1] = std::to_string((int32_t)(unsigned)key);
      } else {
        g.keys[0] = std::to_string(key);
      }
      out->push_back(std::move(g));
    }
    if (!q->sparse_null_row.empty() &&
        q->sparse_null_row[naggs1 - 1] > 0.0) {
      GroupOut g = fill(q->sparse_null_row.data());
      g.key_null[0] = true;               /* the NULL-key group */
      out->push_back(std::move(g));
    }
    return;
  }
  /* concurrent ingest may be interning new dictionary entries; group ids
   * recorded at submit stay valid (dictionaries only grow) but the vector
   * needs the lock for stability */
  std::lock_guard<std::mutex> g(t->mu);
  for (int s = 0; s < q->nslots; s++) {
    const double *row = &q->host_out[(size_t)s * q->out_stride];
    double rowcount = row[2 * q->na_t];
    if ((p.ngroup > 0 || q->join_group) && rowcount == 0.0) continue;
    GroupOut g;
    g.rowcount = rowcount;
    for (int a = 0; a < p.naggs; a++) {
      if (p.ngroup == 0 && !q->join_group) {
        /* keyless MIN/MAX plans run the grouped (pac) layout with 1 slot:
         * counts land at [naggs + a] instead of the keyless [na_t + a] */
        if (a == 0 && getenv("SN_DBG")) {
          fprintf(stderr, "[dbg] pac=%d dev_naggs=%d na_t=%d p.naggs=%d stride=%d row=",
                  q->pac ? 1 : 0, q->dev_naggs, q->na_t, p.naggs, (int)q->out_stride);
          for (size_t i = 0; i < q->out_stride; i++) fprintf(stderr, "%g ", row[i]);
          fprintf(stderr, "\n");
        }
        g.sums[a] = row[a];
        g.counts[a] = q->pac ? row[p.naggs + a] : row[q->na_t + a];
      } else {
        /* grouped: deduped device sweeps; COUNT(*) = rowcount; per-agg
         * counts live at [dev_naggs + di] when the pac layout ran
         * (nullable aggregate inputs), else counts == rowcount */
        int di = q->agg_map[a];
        g.sums[a] = di < 0 ? rowcount : row[di];
        g.counts[a] = (di < 0 || !q->pac) ? rowcount
                                          : row[q->dev_naggs + di];
      }
    }
    if (q->join_group) {
      std::lock_guard<std::mutex> gd(q->join_dim->mu);
      if (p.ngroup >= 1) {
        /* composite: attr-major, fact minor */
        g.keys[0] = q->join_dim->attr_dict[s / q->fact_slots];
        const int f = s % q->fact_slots;
        int c0 = p.group_cols[0];
        if (q->gint[0]) g.keys[1] = std::to_string(q->gmin[0] + f);
        else if (f == q->gnull1) g.key_null[1] = true;
        else g.keys[1] = t->gdict[c0][f];
      } else {
        g.keys[0] = q->join_dim->attr_dict[s];
      }
    } else if (p.ngroup >= 1) {
      int c0 = p.group_cols[0];
      int g1 = s / q->g2cap;
      if (q->gint[0]) g.keys[0] = std::to_string(q->gmin[0] + g1);
      else if (g1 == q->gnull1) g.key_null[0] = true;
      else g.keys[0] = t->gdict[c0][g1];
      if (p.ngroup == 2) {
        int c1 = p.group_cols[1];
        int g2 = s % q->g2cap;
        if (q->gint[1]) g.keys[1] = std::to_string(q->gmin[1] + g2);
        else if (g2 == q->gnull2) g.key_null[1] = true;
        else g.keys[1] = t->gdict[c1][g2];
      }
    }
    out->push_back(std::move(g));
  }
  if (p.ngroup == 0 && !q->join_group && out->empty()) out->push_back(GroupOut());
}

static void finalize_groups(sn_query *q, std::vector<GroupOut> &groups) {
  std::sort(groups.begin(), groups.end(), [](const GroupOut &a, const GroupOut &b) {
    for (int i = 0; i < SN_MAX_GROUPS; i++) {
      if (a.key_null[i] != b.key_null[i]) return !a.key_null[i];
      int c = a.keys[i].compare(b.keys[i]);
      if (c) return c < 0;
    }
    return false;
  });
  q->final_groups = groups;
}

static int32_t result_fill_page(sn_query *q, int64_t offset, sn_result *out) {
  if (!q || !out || offset < 0) return SN_ERR_BADARG;
  int rc = sn_query_wait(q);
  if (rc != SN_OK) return rc;
  if (q->final_groups.empty() && !q->merged) {
    std::vector<GroupOut> groups;
    local_groups(q, &groups);
    finalize_groups(q, groups);
  }
  memset(out, 0, sizeof(*out));
  const sn_plan &p = q->plan;
  out->ngroup = q->join_group ? 1 + p.ngroup : p.ngroup;
  out->naggs = p.naggs;
  int64_t total = (int64_t)q->final_groups.size();
  int64_t start = offset < total ? offset : total;
  out->nrows = (int32_t)std::min((int64_t)SN_MAX_GROUP_SLOTS, total - start);
  out->rows_scanned = q->rows_scanned;
  out->batches_seen = q->batches_seen;
  out->batches_skipped = q->batches_skipped;
  for (int32_t i = 0; i < out->nrows; i++) {
    GroupOut &g = q->final_groups[(size_t)(start + i)];
    for (int k = 0; k < out->ngroup; k++) {
      strncpy(out->keys[i][k], g.keys[k].c_str(), SN_KEY_MAX - 1);
      out->key_is_null[i][k] = g.key_null[k] ? 1 : 0;
    }
    for (int a = 0; a < p.naggs; a++) {
      const sn_agg &ag = p.aggs[a];
      if (ag.kind == SN_AGG_COUNT_STAR) {
        out->vals[i][a] = g.sums[a];
      } else if (ag.kind == SN_AGG_AVG) {
        if (g.counts[a] > 0) out->vals[i][a] = g.sums[a] / g.counts[a];
        else out->val_is_null[i][a] = 1;
      } else {
        if (g.counts[a] > 0) out->vals[i][a] = g.sums[a];
        else out->val_is_null[i][a] = 1;
      }
    }
    out->rows_passed += (int64_t)g.rowcount;
  }
  return SN_OK;
}

extern "C" int32_t sn_query_result(sn_query *q, sn_result *out) {
  return result_fill_page(q, 0, out);
}

/* result paging for group counts beyond SN_MAX_GROUP_SLOTS (the
 * global-atomic grouped path): fills up to one page from `offset`. */
extern "C" int32_t sn_query_result_page(sn_query *q, int64_t offset,
                                        sn_result *out) {
  return result_fill_page(q, offset, out);
}

/* ORDER BY <aggregate value> [DESC] LIMIT k epilogue — the SnappySortExec /
 * TakeOrderedAndProject analogue over the finalized group rows.  The sort
 * key is the RESULT value of aggregate agg_idx (AVG divides, COUNT(*)
 * counts, NULL groups order last); ties keep the key ordering (stable). */
extern "C" int32_t sn_query_order_by(sn_query *q, int32_t agg_idx,
                                     int32_t descending, int64_t k) {
  if (!q) return SN_ERR_BADARG;
  int rc = sn_query_wait(q);
  if (rc != SN_OK) return rc;
  if (q->final_groups.empty() && !q->merged) {
    std::vector<GroupOut> groups;
    local_groups(q, &groups);
    finalize_groups(q, groups);
  }
  const sn_plan &p = q->plan;
  if (agg_idx >= p.naggs) return fail(SN_ERR_BADARG, "agg_idx out of range");
  if (agg_idx >= 0) {
    const sn_agg &ag = p.aggs[agg_idx];
    auto val_of = [&](const GroupOut &g, double *v) -> bool {
      if (ag.kind == SN_AGG_COUNT_STAR) { *v = g.sums[agg_idx]; return true; }
      if (g.counts[agg_idx] <= 0) return false;                 /* NULL */
      *v = ag.kind == SN_AGG_AVG ? g.sums[agg_idx] / g.counts[agg_idx]
                                 : g.sums[agg_idx];
      return true;
    };
    std::stable_sort(q->final_groups.begin(), q->final_groups.end(),
                     [&](const GroupOut &a, const GroupOut &b) {
                       double va, vb;
                       bool ha = val_of(a, &va), hb = val_of(b, &vb);
                       if (ha != hb) return ha;                 /* NULLs last */
                       if (!ha) return false;
                       return descending ? va > vb : va < vb;
                     });
  } else {
    /* restore key order */
    std::vector<GroupOut> groups = q->final_groups;
    finalize_groups(q, groups);
  }
  if (k > 0 && (int64_t)q->final_groups.size() > k)
    q->final_groups.resize((size_t)k);
  return SN_OK;
}

/* total group rows of the finalized result */
extern "C" int64_t sn_query_num_groups(sn_query *q) {
  if (!q) return SN_ERR_BADARG;
  int rc = sn_query_wait(q);
  if (rc != SN_OK) return rc;
  if (q->final_groups.empty() && !q->merged) {
    if (q->sparse) {
      /* count without materializing ~1M string-keyed GroupOuts — the
       * multi-GPU capacity negotiation calls this every step (550 ms vs
       * ~2 ms at 1M groups) */
      const int naggs1 = (q->pac ? 2 : 1) * q->dev_naggs + 1;
      int64_t n = 0;
      for (size_t i = 0; i < q->sparse_n; i++)
        n += q->sparse_rows[i * naggs1 + naggs1 - 1] != 0.0;
      if (!q->sparse_null_row.empty() &&
          q->sparse_null_row[naggs1 - 1] > 0.0)
        n++;
      return n;
    }
    std::vector<GroupOut> groups;
    local_groups(q, &groups);
    finalize_groups(q, groups);
  }
  return (int64_t)q->final_groups.size();
}

static void sn_detach_queries(sn_engine *e) {
  std::lock_guard<std::mutex> g(e->aux_mu);
  for (sn_query *q : e->live_q) q->e = nullptr;
  e->live_q.clear();
}

extern "C" void sn_query_destroy(sn_query *q) {
  if (q && q->e) {
    /* the stream may still reference this query's buffers */
    (void)sn_query_wait(q);
    for (auto &pr : q->owned) q->e->arena.release(pr.first, pr.second);
    std::lock_guard<std::mutex> g(q->e->aux_mu);
    q->e->live_q.erase(q);
  }
  delete q;
}

/* ---- partial exchange ----
 * Grouped blocks carry a caller-chosen slot CAPACITY (the *2 entry points):
 * group counts beyond SN_MAX_GROUP_SLOTS — the big dense-slot and sparse
 * hash-aggregate paths — export by negotiating a capacity across ranks
 * (e.g. allreduce-MAX of sn_query_num_groups) before the collective.  The
 * legacy entry points keep the fixed SN_MAX_GROUP_SLOTS capacity. */
static int64_t partial_bytes_cap(sn_query *q, int32_t cap) {
  if (q->plan.ngroup == 0 && !q->join_group)
    return (int64_t)(2 * q->plan.naggs + 1) * 8;
  return 8 + (int64_t)cap * sizeof(PartialSlot);
}

extern "C" int64_t sn_query_partial_bytes(sn_query *q) {
  if (!q) return SN_ERR_BADARG;
  return partial_bytes_cap(q, SN_MAX_GROUP_SLOTS);
}

extern "C" int64_t sn_query_partial_bytes2(sn_query *q, int32_t cap_slots) {
  if (!q || cap_slots <= 0) return SN_ERR_BADARG;
  return partial_bytes_cap(q, cap_slots);
}

extern "C" int32_t sn_query_partials2(sn_query *q, void *dst,
                                      int32_t dst_is_device, int32_t cap_slots) {
  if (!q || !dst || cap_slots <= 0) return SN_ERR_BADARG;
  int rc = sn_query_wait(q);
  if (rc != SN_OK) return rc;
  const sn_plan &p = q->plan;
  std::vector<uint8_t> block((size_t)partial_bytes_cap(q, cap_slots), 0);
  if (p.ngroup == 0 && !q->join_group) {
    double *o = (double *)block.data();
    const double *row = q->host_out.data();
    for (int a = 0; a < p.naggs; a++) {
      o[a] = row[a];
      o[p.naggs + a] = q->pac ? row[p.naggs + a] : row[q->na_t + a];
    }
    o[2 * p.naggs] = row[2 * q->na_t];
  } else {
    std::vector<GroupOut> groups;
    local_groups(q, &groups);
    if (groups.size() > (size_t)cap_slots)
      return fail(SN_ERR_OVERFLOW,
                  "%zu groups exceed the partial-block capacity %d "
                  "(negotiate a larger capacity via sn_query_partial_bytes2)",
                  groups.size(), cap_slots);
    int32_t n = (int32_t)groups.size();
    memcpy(block.data(), &n, 4);
    memcpy(block.data() + 4, &cap_slots, 4);
    PartialSlot *slots = (PartialSlot *)(block.data() + 8);
    for (int32_t i = 0; i < n; i++) {
      GroupOut &g = groups[i];
      for (int k = 0; k < SN_MAX_GROUPS; k++) {
        strncpy(slots[i].keys[k], g.keys[k].c_str(), SN_KEY_MAX - 1);
        slots[i].key_null[k] = g.key_null[k] ? 1 : 0;
      }
      memcpy(slots[i].sums, g.sums, sizeof(g.sums));
      memcpy(slots[i].counts, g.counts, sizeof(g.counts));
      slots[i].rowcount = g.rowcount;
    }
  }
  if (dst_is_device) {
    HIP_OR_FAIL(hipMemcpy(dst, block.data(), block.size(), hipMemcpyHostToDevice));
  } else {
    memcpy(dst, block.data(), block.size());
  }
  return SN_OK;
}

extern "C" int32_t sn_query_partials(sn_query *q, void *dst, int32_t dst_is_device) {
  return sn_query_partials2(q, dst, dst_is_device, SN_MAX_GROUP_SLOTS);
}

/* Split this shard's grouped partials into `world` same-format blocks by
 * key-hash shard — the key-sharded all-to-all of SURVEY §8(e): rank r keeps
 * only keys hashing to shard r; every other key's partial travels to its
 * owner (the reference's partial->final ShuffleExchange with hash
 * partitioning, SnappyStrategies.scala:560-601).  dst must hold
 * world * sn_query_partial_bytes(q) bytes; block d goes to rank d.  The
 * shard hash is FNV over the key bytes — layout-internal, any deterministic
 * function gives identical results. */
extern "C" int32_t sn_query_partials_sharded2(sn_query *q, int32_t world,
                                              void *dst, int32_t cap_slots) {
  if (!q || !dst || world <= 0 || world > 1024 || cap_slots <= 0)
    return SN_ERR_BADARG;
  const sn_plan &p = q->plan;
  if (p.ngroup == 0 && !q->join_group) return SN_ERR_UNSUPPORTED;
  int rc = sn_query_wait(q);
  if (rc != SN_OK) return rc;
  int64_t bb = partial_bytes_cap(q, cap_slots);
  memset(dst, 0, (size_t)bb * world);
  std::vector<GroupOut> groups;
  local_groups(q, &groups);
  int eff_ngroup = q->join_group ? 1 + p.ngroup : p.ngroup;
  std::vector<int32_t> counts(world, 0);
  for (auto &g : groups) {
    uint64_t h = 1469598103934665603ull;
    for (int k = 0; k < eff_ngroup; k++) {
      if (g.key_null[k]) { h ^= 1; h *= 1099511628211ull; }
      else
        for (char c : g.keys[k]) { h ^= (uint8_t)c; h *= 1099511628211ull; }
      h ^= 0xff; h *= 1099511628211ull;   /* key separator */
    }
    int d = (int)(h % (uint64_t)world);
    uint8_t *bp = (uint8_t *)dst + (int64_t)d * bb;
    int32_t &n = counts[d];
    if (n >= cap_slots)
      return fail(SN_ERR_OVERFLOW,
                  "shard %d exceeds the partial-block capacity %d", d, cap_slots);
    PartialSlot *slots = (PartialSlot *)(bp + 8);
    for (int k = 0; k < SN_MAX_GROUPS; k++) {
      strncpy(slots[n].keys[k], g.keys[k].c_str(), SN_KEY_MAX - 1);
      slots[n].key_null[k] = g.key_null[k] ? 1 : 0;
    }
    memcpy(slots[n].sums, g.sums, sizeof(g.sums));
    memcpy(slots[n].counts, g.counts, sizeof(g.counts));
    slots[n].rowcount = g.rowcount;
    n++;
  }
  for (int d = 0; d < world; d++) {
    uint8_t *bp = (uint8_t *)dst + (int64_t)d * bb;
    memcpy(bp, &counts[d], 4);
    memcpy(bp + 4, &cap_slots, 4);
  }
  return SN_OK;
}

extern "C" int32_t sn_query_partials_sharded(sn_query *q, int32_t world,
                                             void *dst) {
  return sn_query_partials_sharded2(q, world, dst, SN_MAX_GROUP_SLOTS);
}

extern "C" int32_t sn_query_merge(sn_query *q, const void *blocks, int64_t stride,
                                  int32_t n_blocks) {
  if (!q || !blocks || n_blocks <= 0) return SN_ERR_BADARG;
  const sn_plan &p = q->plan;
  std::vector<GroupOut> merged;
  /* op-aware fold of one partial contribution into a GroupOut slot */
  auto fold = [&](GroupOut &g, int a, double s, double c) {
    int k = p.aggs[a].kind;
    if (k == SN_AGG_MIN || k == SN_AGG_MAX) {
      if (c > 0)
        g.sums[a] = g.counts[a] > 0
                        ? (k == SN_AGG_MIN ? std::min(g.sums[a], s)
                                           : std::max(g.sums[a], s))
                        : s;
    } else {
      g.sums[a] += s;
    }
    g.counts[a] += c;
  };
  if (p.ngroup == 0 && !q->join_group) {
    GroupOut g;
    for (int32_t bi = 0; bi < n_blocks; bi++) {
      const double *o = (const double *)((const uint8_t *)blocks + bi * stride);
      for (int a = 0; a < p.naggs; a++) fold(g, a, o[a], o[p.naggs + a]);
      g.rowcount += o[2 * p.naggs];
    }
    merged.push_back(g);
  } else {
    std::map<std::string, GroupOut> bykey;
    for (int32_t bi = 0; bi < n_blocks; bi++) {
      const uint8_t *bp = (const uint8_t *)blocks + bi * stride;
      int32_t n; memcpy(&n, bp, 4);
      const PartialSlot *slots = (const PartialSlot *)(bp + 8);
      int eff_ngroup = q->join_group ? 1 + p.ngroup : p.ngroup;
      for (int32_t i = 0; i < n; i++) {
        std::string key;
        for (int k = 0; k < eff_ngroup; k++) {
          key += slots[i].key_null[k] ? std::string(1, '\x01')
                                      : std::string(slots[i].keys[k]);
          key += '\x00';
        }
        GroupOut &g = bykey[key];
        for (int k = 0; k < eff_ngroup; k++) {
          g.keys[k] = slots[i].keys[k];
          g.key_null[k] = slots[i].key_null[k] != 0;
        }
        for (int a = 0; a < p.naggs; a++)
          fold(g, a, slots[i].sums[a], slots[i].counts[a]);
        g.rowcount += slots[i].rowcount;
      }
    }
    for (auto &kv : bykey) merged.push_back(std::move(kv.second));
  }
  finalize_groups(q, merged);
  q->merged = true;
  return SN_OK;
}
