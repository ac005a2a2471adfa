/*
 * snappy_engine.h — C ABI of the MI355X-native columnar scan/filter/aggregate
 * engine that drops in behind SnappyData's physical-operator surface.
 *
 * This is the JNI-shaped boundary a SnappyData (reference: SnappyDataInc/snappydata
 * @ /root/reference) host JVM would bind to replace the generated
 * WholeStageCodegen scan+aggregate loop.  Each entry point mirrors a reference
 * seam (file:line cites refer to the reference tree):
 *
 *   sn_engine_create / sn_engine_destroy
 *       — engine lifecycle; replaces per-executor codegen context.
 *   sn_table_define
 *       — schema registration; mirrors ColumnFormatRelation table metadata
 *         (core/.../columnar/impl/ColumnFormatRelation.scala) consumed by
 *         PartitionedPhysicalScan.createFromDataSource
 *         (core/.../ExistingPlans.scala:153-209).
 *   sn_batch_put
 *       — accepts one encoded ColumnBatch: exactly the buffers the
 *         ColumnBatchIterator protocol exposes per batch
 *         (core/.../columnar/ColumnBatchIterator.scala:96-163:
 *          next() -> stats blob, getColumnLob(i), getUpdatedColumnDecoder(i)
 *          -> (delta1,delta2), getDeletedColumnDecoder -> delete mask), and the
 *         store seam ExternalStore.storeColumnBatch
 *         (core/.../columnar/ExternalStore.scala:43-45).  Buffer byte formats
 *         are the reference's column encodings (encoders/.../encoding/
 *         ColumnEncoding.scala:37-53 header; Uncompressed.scala;
 *         DictionaryEncoding.scala; RunLengthEncoding.scala;
 *         BooleanBitSetEncoding.scala; ColumnDeleteEncoder.scala:101-126;
 *         ColumnDeltaDecoder.scala:45-58).  The engine copies the buffers into
 *         GPU HBM; the caller may free them after return (the reference's
 *         retain/release contract, ColumnBatchIterator.scala:102-120, becomes
 *         copy-on-put).
 *   sn_query_submit / sn_query_wait / sn_query_result
 *       — replaces plan execution of
 *         ColumnTableScan (core/.../columnar/ColumnTableScan.scala:186-672)
 *         -> Filter -> SnappyHashAggregateExec
 *         (core/.../aggregate/SnappyHashAggregateExec.scala:240-263,337-500)
 *         and driver-side CollectAggregateExec.executeCollect
 *         (core/.../aggregate/CollectAggregateExec.scala:46-91).
 *   sn_query_partials / sn_query_set_partials
 *       — the partial->final aggregation exchange the reference performs with a
 *         Spark ShuffleExchange between partial and final
 *         SnappyHashAggregateExec (requiredChildDistribution,
 *         SnappyHashAggregateExec.scala:161-167).  In the MI355X engine the
 *         exchange is an RCCL collective over xGMI run by the caller (one
 *         process per GPU); these functions export/import the fixed-width
 *         partial-aggregate state through a caller-provided DEVICE buffer.
 *
 * All integers little-endian; all buffers plain (pointer,size) pairs — no
 * torch/JVM types.  Status codes: 0 = OK, negative = error (sn_last_error()
 * returns a message for the calling thread's last failure).
 */
#ifndef SNAPPY_ENGINE_H
#define SNAPPY_ENGINE_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status codes ---- */
#define SN_OK                0
#define SN_ERR_GENERIC      -1
#define SN_ERR_BADARG       -2
#define SN_ERR_NOMEM        -3
#define SN_ERR_BADFORMAT    -4   /* malformed column blob */
#define SN_ERR_UNSUPPORTED  -5   /* plan/encoding shape not supported yet */
#define SN_ERR_NOGPU        -6   /* HIP device unavailable (the engine never
                                    silently falls back to CPU) */
#define SN_ERR_OVERFLOW     -7   /* result/group-count capacity exceeded */

/* ---- scalar types (SQL-side view of a column) ----
 * Mirrors the subset of Spark types on the hot path
 * (TPCHTableSchema.scala:145-163 load path: int/date->int32, double,
 *  char(1)/varchar -> string). */
typedef enum {
  SN_TYPE_INT32  = 0,   /* IntegerType / DateType (days since epoch)  */
  SN_TYPE_INT64  = 1,   /* LongType / TimestampType (micros)          */
  SN_TYPE_DOUBLE = 2,   /* DoubleType                                 */
  SN_TYPE_STRING = 3,   /* StringType (dictionary-encoded on the path)*/
  SN_TYPE_BOOL   = 4,   /* BooleanType                                */
  SN_TYPE_INT16  = 5,   /* ShortType                                  */
  SN_TYPE_INT8   = 6,   /* ByteType                                   */
  SN_TYPE_FLOAT  = 7    /* FloatType                                  */
} sn_type_t;

/* column encodings = reference typeIds (ColumnEncoding.scala:766-773) */
#define SN_ENC_UNCOMPRESSED  0
#define SN_ENC_RUNLENGTH     1
#define SN_ENC_DICTIONARY    2
#define SN_ENC_BIG_DICTIONARY 3
#define SN_ENC_BOOLEAN_BITSET 4

typedef struct {
  const void *data;
  int64_t     len;     /* bytes; data==NULL,len==0 for an absent buffer */
} sn_buf;

typedef struct {
  sn_type_t dtype;
  int32_t   nullable;  /* 0/1 */
} sn_col_schema;

/* ---- engine config (the knobs of io.snappydata.Property,
 *      core/.../Literals.scala:129-159) ---- */
typedef struct {
  int32_t device;            /* HIP device ordinal this engine binds to */
  int64_t column_batch_size; /* bytes; default 24MB (Literals.scala:129-136) */
  int32_t column_max_delta_rows; /* default 10000 (Literals.scala:138-146) */
  int64_t hash_join_size;    /* broadcast-build threshold; default 100MB
                                (Literals.scala:153-159) */
  int32_t n_buckets;         /* partition count for bucket sharding */
  int32_t shard_rank;        /* this process's rank (one process per GPU) */
  int32_t shard_count;       /* world size; batches whose bucket%shard_count
                                != shard_rank are ignored by sn_batch_put */
} sn_config;

typedef struct sn_engine sn_engine;
typedef struct sn_query  sn_query;

/* ---- plan description ----
 * A thin physical plan for the hot-path shapes the reference plans via
 * SnappyStrategies (core/.../SnappyStrategies.scala:340,547-603):
 * scan -> conjunctive range filter -> (grouped|keyless) aggregate
 * [optionally joined to a broadcast dimension table].                   */

#define SN_MAX_PREDS    8
#define SN_MAX_AGGS     12
#define SN_MAX_GROUPS   2     /* group-by columns */
#define SN_MAX_FACTORS  3
#define SN_MAX_GROUP_SLOTS 1024  /* max distinct groups in round-1 path */
#define SN_KEY_MAX      48    /* max bytes of one group-key string */

/* conjunct range predicate on a scanned column:
 * lo OP value OP hi, with each bound optional.
 * Dictionary string columns take equality only (str_eq/str_len): the
 * engine resolves the literal to its global dictionary id and compares
 * ids — the reference's dictionary filter pushdown
 * (ColumnTableScan + DictionaryOptimizedMapAccessor consume the
 * dictionary INDEX instead of the decoded string).                      */
typedef struct {
  int32_t col;          /* scan-table column ordinal */
  int32_t _pad;
  double  lo_d, hi_d;   /* used when the column is DOUBLE/FLOAT */
  int64_t lo_i, hi_i;   /* used when the column is integer/date */
  uint8_t has_lo, has_hi;
  uint8_t lo_strict, hi_strict;  /* 1: strict inequality */
  uint8_t _pad2[4];
  const char *str_eq;   /* string equality literal (STRING cols only);
                           valid for the duration of the submit call */
  int32_t str_len;
  int32_t _pad3;
  /* IN-list membership (the reference's Q12/Q19-class `col IN (...)`
   * filters, compiled into the generated loop by ColumnTableScan):
   * integer/date columns take in_i[in_n]; dictionary string columns take
   * in_s/in_s_len[in_n] (resolved to dictionary ids at submit, like
   * str_eq).  in_n == 0 disables; all pointers valid only during submit.
   * An IN pred may not also carry range bounds. */
  const int64_t *in_i;
  const char *const *in_s;
  const int32_t *in_s_len;
  int32_t in_n;
  int32_t _pad6;
} sn_pred;

/* one multiplicative factor of an aggregate input: (add + mul * col) */
typedef struct {
  int32_t col;
  int32_t _pad;
  double  add;
  double  mul;
} sn_factor;

#define SN_AGG_SUM        0
#define SN_AGG_COUNT_STAR 1
#define SN_AGG_AVG        2
#define SN_AGG_MIN        3
#define SN_AGG_MAX        4

/* aggregate: SUM/AVG/MIN/MAX over a product of factors, or COUNT(*).
 * Null semantics follow Spark Sum/Average/Count and the Min/Max
 * DeclarativeAggregates (SnappyHashAggregateExec.scala:450-500 accumulate
 * rules): a row contributes iff every referenced column is non-null;
 * COUNT(*) counts every surviving row; MIN/MAX of an empty/all-null
 * input group is NULL. */
typedef struct {
  int32_t kind;
  int32_t nfactors;
  sn_factor factors[SN_MAX_FACTORS];
} sn_agg;

#define SN_JOIN_NONE  -1
#define SN_JOIN_SEMI   0   /* fact row survives iff key present in dim */
#define SN_JOIN_GROUP  1   /* additionally GROUP BY the dim attribute;
                              may combine with ONE fact group column
                              (group_cols[0]) — results then carry
                              (attr, fact key) composite keys */

typedef struct {
  int32_t table;        /* handle from sn_table_define */
  int32_t npreds;
  sn_pred preds[SN_MAX_PREDS];
  int32_t ngroup;       /* 0 = keyless aggregate */
  int32_t group_cols[SN_MAX_GROUPS]; /* dictionary string columns, or integer
                           (int16/int32/int64) key columns.  Integer keys
                           with dense stats-derived spans use direct slots;
                           sparse/unbounded integer keys run the
                           open-address hash aggregate (ByteBufferHashMap /
                           SHAMapAccessor analogue) */
  int32_t naggs;
  int32_t _pad;
  sn_agg  aggs[SN_MAX_AGGS];
  /* broadcast-dimension join (HashJoinExec semantics, HashJoinExec.scala:
   * 285-520: per-task ObjectHashSet build from the replicated row-store
   * dimension, cached across tasks via HashedObjectCache :449-470; here the
   * device hash table is built once per dimension and cached on the engine,
   * resident in HBM — the broadcast).  join_dim = SN_JOIN_NONE disables. */
  int32_t join_dim;       /* handle from sn_dim_define */
  int32_t join_fact_col;  /* int32/int64 fact key column */
  int32_t join_mode;      /* SN_JOIN_SEMI | SN_JOIN_GROUP */
  int32_t _pad4;
} sn_plan;

/* ---- results ----
 * Grouped results are returned sorted ascending by key strings (the
 * reference's Q1 "order by l_returnflag, l_linestatus").                */
typedef struct {
  int32_t nrows;        /* number of (group) rows; 1 for keyless */
  int32_t ngroup;
  int32_t naggs;
  int32_t _pad;
  /* row r, key g: NUL-terminated; empty string for SQL NULL key */
  char    keys[SN_MAX_GROUP_SLOTS][SN_MAX_GROUPS][SN_KEY_MAX];
  uint8_t key_is_null[SN_MAX_GROUP_SLOTS][SN_MAX_GROUPS];
  /* row r, agg a: value (AVG already divided; COUNT as exact integer-valued
   * double; SUM of an all-null/empty input group is NULL) */
  double  vals[SN_MAX_GROUP_SLOTS][SN_MAX_AGGS];
  uint8_t val_is_null[SN_MAX_GROUP_SLOTS][SN_MAX_AGGS];
  int64_t rows_scanned;   /* rows examined after batch skipping */
  int64_t rows_passed;    /* rows surviving the filter */
  int64_t batches_seen;   /* SQLMetrics columnBatchesSeen
                             (ColumnTableScan.scala:111-127) */
  int64_t batches_skipped;/* stats-predicate skips (ColumnTableScan.scala:820-963) */
} sn_result;

/* ---- engine lifecycle ---- */
sn_engine *sn_engine_create(const sn_config *cfg);
void       sn_engine_destroy(sn_engine *e);
const char *sn_last_error(void);
/* version/capability probe; returns the gfx arch the .so was built for */
const char *sn_engine_arch(void);

/* ---- schema / data plane ---- */
int32_t sn_table_define(sn_engine *e, const char *name,
                        int32_t ncols, const sn_col_schema *schema);

/* Ingest one encoded column batch (all buffers little-endian, reference
 * formats).  columns[i] = blob for column i (required).  delete_mask and
 * deltas may be absent.  deltas is laid out [ncols][2] (delta1,delta2 per
 * column; ColumnDelta.deltaColumnIndex depths 0..1, ColumnDelta.scala:256-301)
 * — pass NULL when the batch has no update deltas.  stats is the UnsafeRow
 * stats blob (ColumnStatsSchema, ColumnEncoding.scala:1015-1036); pass an
 * empty buf to disable stats-based skipping for this batch. */
int32_t sn_batch_put(sn_engine *e, int32_t table,
                     int64_t uuid, int32_t bucket_id, int32_t num_rows,
                     const sn_buf *columns, const sn_buf *stats,
                     const sn_buf *delete_mask, const sn_buf *deltas);

/* f2 fast ingest (the ColumnBatchCreator rollover with encode + stats
 * offloaded to the GPU): fixed-width NON-NULL columns arrive as RAW value
 * arrays (the Uncompressed encoder for numerics is the identity — the
 * engine uploads the bytes once and computes the ColumnStatsSchema bounds
 * ON DEVICE, async on the ingest stream); other columns arrive as encoded
 * blobs in `encoded`.  Per column exactly one of raw[c].data /
 * encoded[c].data is set.  Queries resolve the pending device bounds with
 * one stream sync. */
int32_t sn_batch_put_raw(sn_engine *e, int32_t table, int64_t uuid,
                         int32_t bucket_id, int32_t num_rows,
                         const sn_buf *raw, const sn_buf *encoded);

/* number of resident batches / rows for a table on this shard */
int64_t sn_table_num_batches(sn_engine *e, int32_t table);
int64_t sn_table_num_rows(sn_engine *e, int32_t table);

/* ---- dimension tables (row-store stand-in for the broadcast join) ----
 * The reference keeps dimension tables in the GemFire row store and builds
 * the probe map per task from region get()s; here the host holds the rows
 * and the engine broadcasts a key -> attribute hash table into HBM. */
int32_t sn_dim_define(sn_engine *e, const char *name);
/* keys must be unique (the dimension's primary key).  attrs: optional
 * per-key attribute strings (payload + lens) for SN_JOIN_GROUP; NULL for
 * semi-join-only dimensions. */
int32_t sn_dim_put(sn_engine *e, int32_t dim, int64_t nkeys,
                   const int64_t *keys, const char *attr_payload,
                   const int32_t *attr_lens);

/* ---- partitioned-partitioned (colocated) join build ----
 * Populate an EMPTY dimension handle from a resident COLUMN TABLE's rows,
 * built on device: the HashJoinExec per-task build of the probe map from
 * the build side (HashJoinExec.scala:285-520) with HashedObjectCache reuse
 * (:449-470).  The reference's partitioned-partitioned joins run
 * bucket-LOCALLY when tables are colocated on the join key (GemFire
 * colocation) — both sides sharded identically means no exchange step, so
 * each shard builds from ITS build-side rows and probes ITS fact rows.
 * key_col: int32/int64 column with UNIQUE values per shard (duplicate keys
 * with conflicting payloads fail with SN_ERR_BADARG); attr_col: optional
 * dictionary-string column whose values become the SN_JOIN_GROUP
 * attributes (-1 = semi-join only).  Rebuild after the build table
 * changes; deleted/patched rows are honored at build time. */
int32_t sn_dim_from_table(sn_engine *e, int32_t dim, int32_t table,
                          int32_t key_col, int32_t attr_col);

/* attach the CURRENT (cumulative) mutation state to an existing batch —
 * the UPDATE/DELETE seam: delete_mask and the per-column delta pairs
 * replace any previous state for that batch; stats (if given) replace the
 * stats row, else stats-based skipping is disabled for the batch.
 * (ColumnDelta.scala:300-301; UpdatedColumnDecoder.scala:69-115) */
int32_t sn_batch_mutate(sn_engine *e, int32_t table, int64_t uuid,
                        int32_t bucket_id, const sn_buf *delete_mask,
                        const sn_buf *deltas, const sn_buf *stats);

/* ---- query plane ---- */
sn_query *sn_query_submit(sn_engine *e, const sn_plan *plan);
/* blocks until device work completes; returns status */
int32_t   sn_query_wait(sn_query *q);
/* fills final (or this shard's partial, before any merge) result */
int32_t   sn_query_result(sn_query *q, sn_result *out);
void      sn_query_destroy(sn_query *q);
/* scan-kernel duration in ms (HIP events on the launch stream); -1 if the
 * query launched nothing */
double    sn_query_kernel_ms(sn_query *q);
/* result paging for group counts beyond SN_MAX_GROUP_SLOTS (dense group
 * spaces up to 2^20 slots run through a global-atomic accumulate path);
 * sn_query_result returns the first page */
int32_t   sn_query_result_page(sn_query *q, int64_t offset, sn_result *out);
int64_t   sn_query_num_groups(sn_query *q);
/* 1 when the query ran a query-compiled (hipRTC) kernel, 0 interpreted */
int32_t   sn_query_used_jit(sn_query *q);
/* ORDER BY <aggregate value> [DESC] LIMIT k epilogue (SnappySortExec /
 * TakeOrderedAndProject semantics, core/.../SnappySortExec.scala): reorders
 * the finalized group rows by aggregate `agg_idx`'s value (NULLs last,
 * ties broken by the key ordering) and truncates to k (k <= 0: no limit).
 * Call after any partial merge; subsequent sn_query_result/`_page` calls
 * return the reordered rows.  agg_idx < 0 restores the key ordering. */
int32_t   sn_query_order_by(sn_query *q, int32_t agg_idx, int32_t descending,
                            int64_t k);

/* ---- multi-GPU partial exchange (caller runs the RCCL collective) ----
 * The caller (one process per GPU, torch.distributed over RCCL/xGMI)
 * exchanges fixed-width partial blocks between the shards' local aggregates
 * and feeds the gathered blocks back for the final merge — mirroring the
 * reference's partial->final ShuffleExchange + CollectAggregateExec.
 *
 * Block layouts (all f64 so the block is directly collective-reducible;
 * counts are exact in f64 up to 2^53):
 *   keyless: [naggs sums][naggs non-null counts][1 row count]
 *            -> identical slot layout on every shard: ncclAllReduce-able.
 *   grouped: [i32 n_slots][i32 capacity] then capacity slots of
 *            { char keys[SN_MAX_GROUPS][SN_KEY_MAX]; u8 key_null[SN_MAX_GROUPS];
 *              u8 pad[6]; f64 sums[naggs]; f64 counts[naggs]; f64 rowcount }
 *            -> self-describing (per-batch dictionaries mean shards need not
 *            agree on slot order): ncclAllGather + key-merge.              */
int64_t sn_query_partial_bytes(sn_query *q);
/* export this shard's partial block; dst_is_device: 1 = HIP device memory */
int32_t sn_query_partials(sn_query *q, void *dst, int32_t dst_is_device);
/* variable-capacity variants for group counts beyond SN_MAX_GROUP_SLOTS
 * (big dense slot spaces and the sparse hash aggregate): the caller
 * negotiates one capacity across ranks (e.g. an allreduce-MAX of
 * sn_query_num_groups) so every block in the collective is the same size —
 * mirroring how the reference's hash-partitioned partial->final exchange
 * sizes its shuffle blocks dynamically (SnappyHashAggregateExec.scala:
 * 161-167).  Block layout is unchanged; capacity rides in the header. */
int64_t sn_query_partial_bytes2(sn_query *q, int32_t cap_slots);
int32_t sn_query_partials2(sn_query *q, void *dst, int32_t dst_is_device,
                           int32_t cap_slots);
int32_t sn_query_partials_sharded2(sn_query *q, int32_t world, void *dst,
                                   int32_t cap_slots);
/* key-sharded split for the grouped all-to-all: write
 * `world` same-format blocks into dst (world * partial_bytes), block d
 * holding only the groups whose key hashes to shard d.  After the
 * all-to-all, each rank merges the world blocks it received (all of which
 * carry only its keys) with sn_query_merge. */
int32_t sn_query_partials_sharded(sn_query *q, int32_t world, void *dst);
/* merge n_blocks partial blocks (stride bytes apart, host memory) into the
 * final result; n_blocks=1 imports an already-all-reduced keyless block */
int32_t sn_query_merge(sn_query *q, const void *blocks, int64_t stride,
                       int32_t n_blocks);

#ifdef __cplusplus
}
#endif
#endif /* SNAPPY_ENGINE_H */
