/*
 * builder.cpp — product-side batch builder + seeded TPC-H lineitem generator.
 *
 * This is the engine's equivalent of the reference's ingest rollover path
 * (ColumnBatchCreator.createAndStoreBatch, ColumnBatchCreator.scala:46-131 +
 * ColumnInsertExec encoding calls): raw row data -> reference-format column
 * blobs (header ColumnEncoding.scala:37-53; uncompressed bodies
 * Uncompressed.scala:32-160; string dictionaries DictionaryEncoding.scala:
 * 85-160 with the int16 -> int32 BigDictionary promotion at Short.MaxValue
 * :313,338) + a ColumnStatsSchema stats UnsafeRow (ColumnEncoding.scala:
 * 1015-1036) -> sn_batch_put.
 *
 * Encoder choice mirrors ColumnEncoding.getColumnEncoder (:838-870):
 * dictionary for strings, uncompressed otherwise.
 *
 * The lineitem generator is the dbgen-equivalent synthetic source mandated by
 * BASELINE.md (no network: seeded, reproducible; distributions chosen to
 * match TPC-H Q6 selectivity ~1.9% — SURVEY.md §8(d)).
 */
#include <cstring>
#include <string>
#include <thread>
#include <vector>
#include <map>
#include <atomic>
#include <cstdio>

#include "../../include/snappy_engine.h"

extern "C" int32_t sn_batch_put(sn_engine *, int32_t, int64_t, int32_t, int32_t,
                                const sn_buf *, const sn_buf *, const sn_buf *,
                                const sn_buf *);
extern "C" int32_t sn_batch_put_raw(sn_engine *, int32_t, int64_t, int32_t,
                                    int32_t, const sn_buf *, const sn_buf *);

/* ---------------- little-endian writers ---------------- */
static inline void put_i32(std::vector<uint8_t> &b, int32_t v) {
  const uint8_t *p = (const uint8_t *)&v; b.insert(b.end(), p, p + 4);
}
static inline void put_i16(std::vector<uint8_t> &b, int16_t v) {
  const uint8_t *p = (const uint8_t *)&v; b.insert(b.end(), p, p + 2);
}

typedef struct { const void *data; const int32_t *str_lens; const uint8_t *valid; } sn_ingest_col;

/* header: typeId + null bitset (ColumnEncoding.scala:37-53) */
static void put_header(std::vector<uint8_t> &out, int32_t type_id,
                       const uint8_t *valid, int32_t count) {
  put_i32(out, type_id);
  int32_t nnull = 0;
  if (valid) for (int32_t i = 0; i < count; i++) nnull += !valid[i];
  if (!nnull) { put_i32(out, 0); return; }
  int32_t words = (count + 63) >> 6;
  put_i32(out, words * 8);
  size_t base = out.size();
  out.resize(base + (size_t)words * 8, 0);
  for (int32_t i = 0; i < count; i++)
    if (!valid[i]) out[base + (i >> 6) * 8 + ((i & 63) >> 3)] |= (uint8_t)(1u << (i & 7));
}

/* encode one column (uncompressed numerics / dictionary strings) */
static int encode_column(sn_type_t dtype, const sn_ingest_col &col,
                         int32_t count, std::vector<uint8_t> &out) {
  const uint8_t *v8 = (const uint8_t *)col.data;
  const uint8_t *valid = col.valid;
  if (dtype == SN_TYPE_STRING) {
    /* two-pass dictionary build in first-occurrence order; promote to
     * BigDictionary when entries reach Short.MaxValue */
    std::map<std::string, int32_t> didx;
    std::vector<std::string> dict;
    std::vector<int32_t> idx(count, -1);
    int64_t so = 0;
    for (int32_t i = 0; i < count; i++) {
      int32_t L = col.str_lens[i];
      if (valid && !valid[i]) { so += L; continue; }
      std::string s((const char *)v8 + so, (size_t)L);
      so += L;
      auto it = didx.find(s);
      if (it == didx.end()) {
        idx[i] = (int32_t)dict.size();
        didx.emplace(std::move(s), idx[i]);
        dict.push_back(std::string((const char *)v8 + so - L, (size_t)L));
      } else idx[i] = it->second;
    }
    bool big = dict.size() >= 32767;
    put_header(out, big ? SN_ENC_BIG_DICTIONARY : SN_ENC_DICTIONARY, valid, count);
    put_i32(out, (int32_t)dict.size());
    for (auto &s : dict) { put_i32(out, (int32_t)s.size());
      out.insert(out.end(), s.begin(), s.end()); }
    for (int32_t i = 0; i < count; i++) {
      if (idx[i] < 0) continue;          /* writeIsNull writes no index */
      if (big) put_i32(out, idx[i]); else put_i16(out, (int16_t)idx[i]);
    }
    return SN_OK;
  }
  put_header(out, SN_ENC_UNCOMPRESSED, valid, count);
  int w;
  switch (dtype) {
    case SN_TYPE_DOUBLE: case SN_TYPE_INT64: w = 8; break;
    case SN_TYPE_INT32: case SN_TYPE_FLOAT: w = 4; break;
    case SN_TYPE_INT16: w = 2; break;
    case SN_TYPE_INT8: case SN_TYPE_BOOL: w = 1; break;
    default: return SN_ERR_UNSUPPORTED;
  }
  if (!valid) {
    out.insert(out.end(), v8, v8 + (size_t)count * w);
  } else {
    for (int32_t i = 0; i < count; i++)
      if (valid[i]) out.insert(out.end(), v8 + (size_t)i * w, v8 + (size_t)(i + 1) * w);
  }
  return SN_OK;
}

/* stats row writer (UnsafeRow: ColumnStatsSchema) */
static void encode_stats(const std::vector<sn_type_t> &dtypes,
                         int32_t batch_count_signed,
                         const std::vector<double> &lo_d, const std::vector<double> &hi_d,
                         const std::vector<int64_t> &lo_i, const std::vector<int64_t> &hi_i,
                         const std::vector<int32_t> &ncount,
                         const std::vector<uint8_t> &has_bounds,
                         std::vector<uint8_t> &out) {
  int nc = (int)dtypes.size();
  int32_t num_fields = nc * 3 + 1;
  int32_t nwords = (num_fields + 63) >> 6;
  out.assign((size_t)nwords * 8 + (size_t)num_fields * 8, 0);
  uint8_t *bits = out.data();
  uint8_t *slots = out.data() + (size_t)nwords * 8;
  auto set_bit = [&](int f) { bits[(f >> 6) * 8 + ((f & 63) >> 3)] |= (uint8_t)(1u << (f & 7)); };
  memcpy(slots, &batch_count_signed, 4);
  for (int c = 0; c < nc; c++) {
    int f_lo = 1 + c * 3, f_hi = 2 + c * 3, f_nc = 3 + c * 3;
    memcpy(slots + (size_t)f_nc * 8, &ncount[c], 4);
    if (!has_bounds[c]) { set_bit(f_lo); set_bit(f_hi); continue; }
    switch (dtypes[c]) {
      case SN_TYPE_DOUBLE: case SN_TYPE_FLOAT:
        memcpy(slots + (size_t)f_lo * 8, &lo_d[c], 8);
        memcpy(slots + (size_t)f_hi * 8, &hi_d[c], 8); break;
      case SN_TYPE_INT64:
        memcpy(slots + (size_t)f_lo * 8, &lo_i[c], 8);
        memcpy(slots + (size_t)f_hi * 8, &hi_i[c], 8); break;
      default: {
        int32_t lo = (int32_t)lo_i[c], hi = (int32_t)hi_i[c];
        memcpy(slots + (size_t)f_lo * 8, &lo, 4);
        memcpy(slots + (size_t)f_hi * 8, &hi, 4);
      }
    }
  }
}

/* schema/shard accessors implemented in engine.cpp */
struct TableInfoProbe { int32_t ncols; sn_type_t dtypes[64]; uint8_t nullable[64]; };
extern "C" int32_t sn_table_schema(sn_engine *e, int32_t table, TableInfoProbe *out);
extern "C" void sn_engine_shard(sn_engine *e, int32_t *rank, int32_t *count);

/* ingest raw row data: split into batches, encode, put.
 * Returns rows ingested (on this shard) or negative error. */
extern "C" int64_t sn_ingest_columns(sn_engine *e, int32_t table, int64_t nrows,
                                     const sn_ingest_col *cols, int32_t batch_rows,
                                     int32_t first_bucket) {
  TableInfoProbe ti;
  if (sn_table_schema(e, table, &ti) != SN_OK) return SN_ERR_BADARG;
  if (batch_rows <= 0) batch_rows = 200000;
  int nc = ti.ncols;
  int64_t put_rows = 0;
  std::vector<int64_t> str_off(nc, 0);
  for (int64_t s = 0, bi = 0; s < nrows; s += batch_rows, bi++) {
    int32_t n = (int32_t)std::min<int64_t>(batch_rows, nrows - s);
    std::vector<std::vector<uint8_t>> blobs(nc);
    std::vector<sn_buf> raw(nc), enc(nc);
    for (int c = 0; c < nc; c++) {
      raw[c].data = nullptr; raw[c].len = 0;
      enc[c].data = nullptr; enc[c].len = 0;
      sn_ingest_col view = cols[c];
      const uint8_t *base = (const uint8_t *)cols[c].data;
      if (cols[c].valid) view.valid = cols[c].valid + s;
      if (ti.dtypes[c] == SN_TYPE_STRING) {
        view.data = base + str_off[c];
        view.str_lens = cols[c].str_lens + s;
        int64_t bytes = 0;
        for (int32_t i = 0; i < n; i++) bytes += cols[c].str_lens[s + i];
        str_off[c] += bytes;
      } else {
        int w = (ti.dtypes[c] == SN_TYPE_DOUBLE || ti.dtypes[c] == SN_TYPE_INT64) ? 8 :
                (ti.dtypes[c] == SN_TYPE_INT16) ? 2 :
                (ti.dtypes[c] == SN_TYPE_INT8 || ti.dtypes[c] == SN_TYPE_BOOL) ? 1 : 4;
        view.data = base + (int64_t)s * w;
        bool rawable = !view.valid &&
            (ti.dtypes[c] == SN_TYPE_DOUBLE || ti.dtypes[c] == SN_TYPE_FLOAT ||
             ti.dtypes[c] == SN_TYPE_INT64 || ti.dtypes[c] == SN_TYPE_INT32 ||
             ti.dtypes[c] == SN_TYPE_INT16);
        if (rawable) {
          /* f2 fast path: the Uncompressed encoder for non-null numerics is
           * the identity — hand the raw slice to the engine, which uploads
           * it once and computes the stats bounds ON DEVICE */
          raw[c].data = view.data;
          raw[c].len = (int64_t)n * w;
          continue;
        }
      }
      int rc = encode_column(ti.dtypes[c], view, n, blobs[c]);
      if (rc != SN_OK) return rc;
      enc[c].data = blobs[c].data();
      enc[c].len = (int64_t)blobs[c].size();
    }
    int32_t rc = sn_batch_put_raw(e, table, bi, first_bucket + (int32_t)bi, n,
                                  raw.data(), enc.data());
    if (rc != SN_OK) return rc;
    put_rows += n;
  }
  return put_rows;
}

/* ---------------- seeded lineitem generator ---------------- */
static inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97f4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

static int days_from_civil(int y, int m, int d) {
  y -= m <= 2;
  int era = (y >= 0 ? y : y - 399) / 400;
  unsigned yoe = (unsigned)(y - era * 400);
  unsigned doy = (153u * (unsigned)(m + (m > 2 ? -3 : 9)) + 2) / 5 + (unsigned)d - 1;
  unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return era * 146097 + (int)doe - 719468;
}

extern "C" void sn_gen_lineitem_arrays(int64_t start_row, int32_t n, int64_t seed,
                                       double *qty, double *ep, double *disc,
                                       double *tax, uint8_t *rf, uint8_t *ls,
                                       int32_t *ship) {
  const int ship_lo = days_from_civil(1992, 1, 1);
  const int ship_hi = days_from_civil(1998, 12, 1);
  static const char RF[3] = { 'A', 'N', 'R' };
  static const char LS[2] = { 'F', 'O' };
  for (int32_t i = 0; i < n; i++) {
    uint64_t r = (uint64_t)(start_row + i);
    uint64_t a = splitmix64(r * 2 + 1 + (uint64_t)seed * 0x100000001ull);
    uint64_t b = splitmix64(r * 2 + 2 + (uint64_t)seed * 0x100000001ull);
    qty[i] = (double)(1 + (a % 50));
    ep[i] = (double)(90000 + ((a >> 8) % 10410000)) / 100.0;
    disc[i] = (double)((a >> 32) % 11) / 100.0;
    tax[i] = (double)((a >> 40) % 9) / 100.0;
    rf[i] = (uint8_t)RF[(b >> 3) % 3];
    ls[i] = (uint8_t)LS[(b >> 5) % 2];
    ship[i] = ship_lo + (int32_t)((b >> 16) % (uint64_t)(ship_hi - ship_lo));
  }
}

/* generate + ingest `total_rows` of lineitem into `table` (7-column hot-path
 * projection: qty, ep, disc, tax, returnflag, linestatus, shipdate).
 * Batches are deterministic in (seed, batch index) so every shard layout
 * produces identical data; a batch's bucket = its index, and sn_batch_put
 * keeps only buckets owned by this shard.  Multithreaded generation+encode.
 * Returns rows resident on this shard. */
extern "C" int64_t sn_datagen_lineitem(sn_engine *e, int32_t table,
                                       int64_t total_rows, int64_t seed,
                                       int32_t batch_rows, int32_t nthreads) {
  TableInfoProbe ti;
  if (sn_table_schema(e, table, &ti) != SN_OK) return SN_ERR_BADARG;
  if (ti.ncols != 7) return SN_ERR_BADARG;
  if (batch_rows <= 0) batch_rows = 600000;   /* ~24 MB across 7 columns */
  if (nthreads <= 0) nthreads = (int32_t)std::thread::hardware_concurrency();
  /* oversubscription measured counterproductive (allocator/copy contention) */
  if (nthreads > 64) nthreads = 64;
  int64_t nbatches = (total_rows + batch_rows - 1) / batch_rows;
  std::atomic<int64_t> next(0), put_rows(0);
  std::atomic<int32_t> err(SN_OK);
  std::vector<sn_type_t> dtypes(ti.dtypes, ti.dtypes + 7);

  /* which buckets this shard owns is decided inside sn_batch_put; we probe
   * it cheaply here to skip generating foreign batches */
  auto worker = [&]() {
    std::vector<double> qty(batch_rows), ep(batch_rows), disc(batch_rows), tax(batch_rows);
    std::vector<uint8_t> rf(batch_rows), ls(batch_rows);
    std::vector<int32_t> ship(batch_rows);
    std::vector<int32_t> len1(batch_rows, 1);
    int32_t sh_rank, sh_count;
    sn_engine_shard(e, &sh_rank, &sh_count);
    while (true) {
      int64_t bi = next.fetch_add(1);
      if (bi >= nbatches || err.load() != SN_OK) break;
      /* skip batches another shard owns (bucket = batch index) */
      if (sh_count > 1 && (int32_t)(bi % sh_count) != sh_rank) continue;
      int32_t n = (int32_t)std::min<int64_t>(batch_rows, total_rows - bi * batch_rows);
      int64_t rows_before = bi * batch_rows;
      sn_gen_lineitem_arrays(rows_before, n, seed, qty.data(), ep.data(),
                             disc.data(), tax.data(), rf.data(), ls.data(),
                             ship.data());
      /* f2 fast path: numerics ship RAW (encode = identity, stats on
       * device); only the two 1-char dictionary string columns encode */
      std::vector<std::vector<uint8_t>> blobs(2);
      sn_ingest_col c;
      c.valid = nullptr;
      c.str_lens = len1.data();
      c.data = rf.data(); encode_column(SN_TYPE_STRING, c, n, blobs[0]);
      c.data = ls.data(); encode_column(SN_TYPE_STRING, c, n, blobs[1]);
      sn_buf raw[7], enc[7];
      for (int i = 0; i < 7; i++) {
        raw[i].data = nullptr; raw[i].len = 0;
        enc[i].data = nullptr; enc[i].len = 0;
      }
      raw[0] = { qty.data(), (int64_t)n * 8 };
      raw[1] = { ep.data(), (int64_t)n * 8 };
      raw[2] = { disc.data(), (int64_t)n * 8 };
      raw[3] = { tax.data(), (int64_t)n * 8 };
      enc[4] = { blobs[0].data(), (int64_t)blobs[0].size() };
      enc[5] = { blobs[1].data(), (int64_t)blobs[1].size() };
      raw[6] = { ship.data(), (int64_t)n * 4 };
      int32_t rc = sn_batch_put_raw(e, table, bi, (int32_t)bi, n, raw, enc);
      if (rc != SN_OK) { err.store(rc); break; }
      put_rows += n;
    }
  };
  std::vector<std::thread> ths;
  for (int i = 0; i < nthreads; i++) ths.emplace_back(worker);
  for (auto &th : ths) th.join();
  if (err.load() != SN_OK) return err.load();
  return put_rows.load();
}

/* ---------------- product-side mutation encoders ----------------
 * Mirrors the reference's write-side delta/delete encoders:
 *   delete mask [int32 0][int32 numBaseRows][int32 numPositions][sorted
 *   int32 positions] — ColumnDeleteEncoder.createFinalBuffer
 *   (ColumnDeleteEncoder.scala:101-126);
 *   update delta: standard blob header (null bitset over DELTA entries) +
 *   [numBaseRows][numPositions][positions][pad8][uncompressed body] —
 *   ColumnDeltaEncoder header doc (ColumnDeltaEncoder.scala:40-80).       */
extern "C" int64_t sn_encode_delete_mask(const int32_t *pos, int32_t n,
                                         int32_t num_base_rows, uint8_t *out,
                                         int64_t cap) {
  int64_t need = 12 + (int64_t)n * 4;
  if (cap < need) return SN_ERR_NOMEM;
  int32_t zero = 0;
  memcpy(out, &zero, 4);
  memcpy(out + 4, &num_base_rows, 4);
  memcpy(out + 8, &n, 4);
  memcpy(out + 12, pos, (size_t)n * 4);
  return need;
}

extern "C" int64_t sn_encode_update_delta(int32_t dtype, const int32_t *pos,
                                          int32_t n, int32_t num_base_rows,
                                          const void *values,
                                          const int32_t *str_lens,
                                          const uint8_t *valid, uint8_t *out,
                                          int64_t cap) {
  sn_ingest_col col;
  col.data = values; col.str_lens = str_lens; col.valid = valid;
  std::vector<uint8_t> body;
  int rc = encode_column((sn_type_t)dtype, col, n, body);
  if (rc != SN_OK) return rc;
  /* splice positions between the null header and the encoding body */
  int32_t null_bytes;
  memcpy(&null_bytes, body.data() + 4, 4);
  int64_t hdr = 8 + null_bytes;
  int64_t off = hdr + 8 + (int64_t)n * 4;
  int64_t pad = ((off + 7) & ~7ll) - off;
  int64_t need = (int64_t)body.size() + 8 + (int64_t)n * 4 + pad;
  if (cap < need) return SN_ERR_NOMEM;
  memcpy(out, body.data(), (size_t)hdr);
  memcpy(out + hdr, &num_base_rows, 4);
  memcpy(out + hdr + 4, &n, 4);
  memcpy(out + hdr + 8, pos, (size_t)n * 4);
  memset(out + off, 0, (size_t)pad);
  memcpy(out + off + pad, body.data() + hdr, body.size() - (size_t)hdr);
  return need;
}

/* expose the column encoder itself (ColumnEncoder mirror) so external hosts
 * can build reference-format blobs without the engine's batching policy */
extern "C" int64_t sn_encode_column(int32_t dtype, const void *data,
                                    const int32_t *str_lens,
                                    const uint8_t *valid, int32_t count,
                                    uint8_t *out, int64_t cap) {
  sn_ingest_col col;
  col.data = data; col.str_lens = str_lens; col.valid = valid;
  std::vector<uint8_t> body;
  int rc = encode_column((sn_type_t)dtype, col, count, body);
  if (rc != SN_OK) return rc;
  if ((int64_t)body.size() > cap) return SN_ERR_NOMEM;
  memcpy(out, body.data(), body.size());
  return (int64_t)body.size();
}
