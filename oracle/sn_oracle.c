/*
 * sn_oracle.c — CPU oracle for the MI355X-native SnappyData columnar engine.
 *
 * TEST INFRASTRUCTURE ONLY.  This file is a from-scratch C restatement of the
 * reference's (SnappyDataInc/snappydata @ /root/reference) column byte formats
 * and generated scan->filter->hash-aggregate loop semantics.  It exists to be
 * the parity checker and the reported CPU baseline.  Only tests/,
 * __graft_entry__.smoke() and bench.py's cpu_baseline leg may link, load or
 * call anything in this directory.  The product path (snappydata_amd/) must
 * never route through it.
 *
 * Pinning: the reference cannot be compiled here (Scala 2.11/JVM, no javac,
 * store/spark submodules un-vendored).  This oracle is therefore pinned
 * against the reference's own end-to-end golden vectors:
 *   tests/common/src/main/resources/TPCH/RESULT/Snappy_1.out / Snappy_6.out
 * computed from the bundled TPCH/lineitem.tbl (30,201 rows), validated by the
 * reference's TPCHDUnitTest.scala:57-71,643-713 — see tests/test_oracle_tpch.py.
 * Encoder byte-level layout has NO golden bytes in the reference (only
 * round-trip property tests, ColumnEncodersTest.scala:26-33); this oracle pins
 * the layout itself per the file:line specs below, with new vectors committed
 * under tests/golden/.
 *
 * Format restatements (reference file:line):
 *  - blob header [int32 typeId][int32 numNullBytes][null bitset][body],
 *    little-endian: encoders/.../encoding/ColumnEncoding.scala:37-53 (doc),
 *    :1080-1092 (read+assert), :1317 (write); typeIds :766-773.
 *  - null bitset = 64-bit words, set bit == NULL at that ordinal; sequential
 *    next-null-position decoding: NullableDecoder ColumnEncoding.scala:1069-1135.
 *  - Uncompressed body: fixed width little-endian array indexed by
 *    nonNullPosition; var-width [int32 size][bytes] sequential:
 *    Uncompressed.scala:32-160.
 *  - RunLength body: [value][int32 run-length] repeated, run values cumulative
 *    to a position: RunLengthEncoding.scala:88-173 (decoder only; the vendored
 *    tree has NO RunLength encoder — ColumnEncoding.getColumnEncoder TODO at
 *    :840 — so RLE layout is pinned by this oracle's own vectors; the byte-
 *    width case in the reference decoder (advance 3 but read int32 at +1,
 *    RunLengthEncoding.scala:99-110) is unreachable/broken and is not
 *    supported here: short=6B, int=8B, long=12B, string=4+len+4 per run).
 *  - Dictionary body: [int32 numElems][entries][int16 indexes] (typeId 2) or
 *    [int32 indexes] (BigDictionary typeId 3); string entry [int32 len][utf8];
 *    null == index numElems: DictionaryEncoding.scala:85-160.
 *  - BooleanBitSet body: plain 64-bit-word bitset of values:
 *    BooleanBitSetEncoding.scala:27-79.
 *  - delete mask [int32 0][int32 numBaseRows][int32 numPositions][sorted
 *    int32 positions]: ColumnDeleteEncoder.scala:101-126, decoder
 *    ColumnDeleteDecoder.scala:24-55.
 *  - update delta: standard header, then [int32 numBaseRows][int32
 *    numPositions][sorted int32 batch positions][pad to 8][encoded values
 *    body]; the delta's null bitset indexes DELTA entries: ColumnDeltaDecoder.
 *    scala:31-60 (initialize :45-58); 2-deep merge where delta1 overrides
 *    delta2 overrides base: UpdatedColumnDecoder.scala:69-115.
 *  - stats row: Spark UnsafeRow of [batchCount:int][per col lowerBound,
 *    upperBound,nullCount] (3/col+1): ColumnStatsSchema
 *    ColumnEncoding.scala:1015-1036; UnsafeRow layout [8B-aligned null words]
 *    [numFields x 8B slots][var tail] via SharedUtils.toUnsafeRow
 *    (encoders/.../collection/SharedUtils.scala:68-79); negative batchCount
 *    => batch has update deltas (ColumnTableScan.scala:524-528).
 *  - scan loop semantics (row order, per-row delete check, null tracking,
 *    delta merge): ColumnTableScan.scala:186-672 generated loop.
 *  - aggregate semantics (Spark Sum/Average/Count update rules, partial->
 *    final): SnappyHashAggregateExec.scala:337-500.
 *  - stats-predicate batch skip: ColumnTableScan.generateStatPredicate
 *    :820-963 (bound checks only; conservative here: batches with deltas or
 *    deletes are never skipped).
 */
#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <stdio.h>
#include <math.h>
#include <dlfcn.h>

#include "../include/snappy_engine.h"

#ifdef _OPENMP
#include <omp.h>
#endif

#define SNO_EXPORT __attribute__((visibility("default")))

/* ---------------- little-endian primitive IO (host is LE x86) ----------- */
static inline int32_t rd_i32(const uint8_t *p) { int32_t v; memcpy(&v, p, 4); return v; }
static inline int64_t rd_i64(const uint8_t *p) { int64_t v; memcpy(&v, p, 8); return v; }
static inline int16_t rd_i16(const uint8_t *p) { int16_t v; memcpy(&v, p, 2); return v; }
static inline double  rd_f64(const uint8_t *p) { double  v; memcpy(&v, p, 8); return v; }
static inline float   rd_f32(const uint8_t *p) { float   v; memcpy(&v, p, 4); return v; }
static inline void wr_i32(uint8_t *p, int32_t v) { memcpy(p, &v, 4); }
static inline void wr_i64(uint8_t *p, int64_t v) { memcpy(p, &v, 8); }
static inline void wr_i16(uint8_t *p, int16_t v) { memcpy(p, &v, 2); }
static inline void wr_f64(uint8_t *p, double v)  { memcpy(p, &v, 8); }
static inline void wr_f32(uint8_t *p, float v)   { memcpy(p, &v, 4); }

/* ---------------- 64-bit-word bitset (encoders/.../encoding/BitSet.scala) */
static inline int bitset_get(const uint8_t *words, int32_t pos) {
  uint64_t w; memcpy(&w, words + ((pos >> 6) << 3), 8);
  return (int)((w >> (pos & 63)) & 1u);
}
/* next set bit at >= from, over num_words words; INT32_MAX when none */
static int32_t bitset_next_set(const uint8_t *words, int32_t num_words, int32_t from) {
  if (from < 0) from = 0;
  int32_t wi = from >> 6;
  if (wi >= num_words) return INT32_MAX;
  uint64_t w; memcpy(&w, words + ((size_t)wi << 3), 8);
  w &= ~0ULL << (from & 63);
  while (1) {
    if (w) return (wi << 6) + (int32_t)__builtin_ctzll(w);
    if (++wi >= num_words) return INT32_MAX;
    memcpy(&w, words + ((size_t)wi << 3), 8);
  }
}
static inline void bitset_set(uint8_t *words, int32_t pos) {
  uint64_t w; memcpy(&w, words + ((pos >> 6) << 3), 8);
  w |= 1ULL << (pos & 63);
  memcpy(words + ((pos >> 6) << 3), &w, 8);
}

/* ---------------- dtype widths ---------------- */
static int type_width(int32_t dtype) {
  switch (dtype) {
    case SN_TYPE_INT32: return 4;
    case SN_TYPE_INT64: return 8;
    case SN_TYPE_DOUBLE: return 8;
    case SN_TYPE_BOOL: return 1;
    case SN_TYPE_INT16: return 2;
    case SN_TYPE_INT8: return 1;
    case SN_TYPE_FLOAT: return 4;
    default: return 0; /* string: variable */
  }
}

/* =======================================================================
 * Column blob decoder (sequential, ordinal-driven; restates ColumnDecoder
 * construction order ColumnEncoding.scala:64-74: typeId -> nulls -> delta
 * positions -> encoding body)
 * ======================================================================= */
typedef struct {
  const uint8_t *blob;
  int64_t len;
  int32_t dtype;
  int32_t type_id;           /* encoding */
  /* nulls */
  const uint8_t *null_words; /* NULL if none */
  int32_t num_null_words;
  int32_t next_null;         /* next null ordinal (INT32_MAX none) */
  int32_t nulls_before;      /* count of nulls at ordinals < current */
  /* body */
  const uint8_t *body;       /* base cursor for fixed-width / index array */
  /* dictionary */
  const uint8_t *dict_entries; /* start of entries region */
  int32_t dict_n;              /* numElements */
  const int32_t *dict_offsets; /* for strings: offset of entry i payload */
  int32_t *dict_offsets_own;
  /* RLE state */
  int64_t rle_cursor;
  int32_t rle_end_pos;       /* runLengthEndPosition */
  int64_t rle_val_i;
  const uint8_t *rle_str; int32_t rle_str_len;
  /* var-width (uncompressed string) state */
  int64_t var_cursor;
  int32_t var_last_pos;
} sno_dec;

static int dec_init(sno_dec *d, int32_t dtype, const uint8_t *blob, int64_t len,
                    int is_delta, const uint8_t **delta_pos_out, int32_t *delta_npos_out) {
  memset(d, 0, sizeof(*d));
  if (!blob || len < 8) return SN_ERR_BADFORMAT;
  d->blob = blob; d->len = len; d->dtype = dtype;
  d->type_id = rd_i32(blob);
  if (d->type_id < 0 || d->type_id > 4) return SN_ERR_BADFORMAT;
  int64_t cur = 4;
  int32_t null_bytes = rd_i32(blob + cur); cur += 4;
  if (null_bytes < 0 || (null_bytes & 7) || cur + null_bytes > len) return SN_ERR_BADFORMAT;
  if (null_bytes > 0) {
    d->null_words = blob + cur;
    d->num_null_words = null_bytes >> 3;
    cur += null_bytes;
    d->next_null = bitset_next_set(d->null_words, d->num_null_words, 0);
  } else {
    d->next_null = INT32_MAX;
  }
  if (is_delta) {
    /* ColumnDeltaDecoder.initialize (ColumnDeltaDecoder.scala:45-58):
     * [numBaseRows][numPositions][positions], data rounded up to 8 */
    if (cur + 8 > len) return SN_ERR_BADFORMAT;
    int32_t npos = rd_i32(blob + cur + 4);
    if (npos < 0) return SN_ERR_BADFORMAT;
    *delta_pos_out = blob + cur + 8;
    *delta_npos_out = npos;
    cur = cur + 8 + ((int64_t)npos << 2);
    cur = (cur + 7) & ~7LL;
    if (cur > len) return SN_ERR_BADFORMAT;
  }
  switch (d->type_id) {
    case SN_ENC_UNCOMPRESSED:
      d->body = blob + cur;
      d->var_cursor = cur;
      d->var_last_pos = -1;
      break;
    case SN_ENC_RUNLENGTH:
      d->rle_cursor = cur;
      d->rle_end_pos = -1;
      break;
    case SN_ENC_DICTIONARY:
    case SN_ENC_BIG_DICTIONARY: {
      if (cur + 4 > len) return SN_ERR_BADFORMAT;
      int32_t n = rd_i32(blob + cur); cur += 4;
      if (n < 0) return SN_ERR_BADFORMAT;
      d->dict_n = n;
      d->dict_entries = blob + cur;
      if (dtype == SN_TYPE_STRING) {
        d->dict_offsets_own = (int32_t *)malloc(sizeof(int32_t) * (size_t)(n + 1));
        if (!d->dict_offsets_own) return SN_ERR_NOMEM;
        int64_t c = cur;
        for (int32_t i = 0; i < n; i++) {
          if (c + 4 > len) return SN_ERR_BADFORMAT;
          int32_t sz = rd_i32(blob + c);
          d->dict_offsets_own[i] = (int32_t)(c - cur);
          c += 4 + sz;
          if (sz < 0 || c > len) return SN_ERR_BADFORMAT;
        }
        d->dict_offsets_own[n] = (int32_t)(c - cur);
        d->dict_offsets = d->dict_offsets_own;
        cur = c;
      } else if (dtype == SN_TYPE_INT32) {
        cur += (int64_t)n << 2;
      } else if (dtype == SN_TYPE_INT64) {
        cur += (int64_t)n << 3;
      } else return SN_ERR_UNSUPPORTED;
      if (cur > len) return SN_ERR_BADFORMAT;
      d->body = blob + cur; /* index array */
      break;
    }
    case SN_ENC_BOOLEAN_BITSET:
      if (dtype != SN_TYPE_BOOL) return SN_ERR_BADFORMAT;
      d->body = blob + cur;
      break;
  }
  return SN_OK;
}

static void dec_free(sno_dec *d) { free(d->dict_offsets_own); d->dict_offsets_own = NULL; }

/* sequential null check at ordinal `ord` (must be called for every ordinal in
 * ascending order).  Returns 1 if NULL (and advances null cursor). */
static inline int dec_null_advance(sno_dec *d, int32_t ord) {
  if (ord == d->next_null) {
    d->nulls_before++;
    d->next_null = bitset_next_set(d->null_words, d->num_null_words, ord + 1);
    return 1;
  }
  return 0;
}

/* random-access null check (used by delta decoders, UpdatedColumnDecoder path) */
static inline int dec_is_null_at(const sno_dec *d, int32_t pos) {
  if (!d->null_words) return 0;
  if ((pos >> 6) >= d->num_null_words) return 0;
  return bitset_get(d->null_words, pos);
}

/* read numeric value as int64 at nonNullPosition (integer dtypes) */
static int64_t dec_read_i64(sno_dec *d, int32_t nnp) {
  switch (d->type_id) {
    case SN_ENC_UNCOMPRESSED:
      switch (d->dtype) {
        case SN_TYPE_INT32: return rd_i32(d->body + ((int64_t)nnp << 2));
        case SN_TYPE_INT64: return rd_i64(d->body + ((int64_t)nnp << 3));
        case SN_TYPE_INT16: return rd_i16(d->body + ((int64_t)nnp << 1));
        case SN_TYPE_INT8:  return (int8_t)d->body[nnp];
        case SN_TYPE_BOOL:  return d->body[nnp] == 1;
        default: return 0;
      }
    case SN_ENC_RUNLENGTH: {
      if (d->rle_end_pos >= nnp) return d->rle_val_i;
      int w = (d->dtype == SN_TYPE_INT16) ? 2 : (d->dtype == SN_TYPE_INT32) ? 4 : 8;
      do {
        const uint8_t *p = d->blob + d->rle_cursor;
        if (w == 2) d->rle_val_i = rd_i16(p);
        else if (w == 4) d->rle_val_i = rd_i32(p);
        else d->rle_val_i = rd_i64(p);
        d->rle_end_pos += rd_i32(p + w);
        d->rle_cursor += w + 4;
      } while (d->rle_end_pos < nnp);
      return d->rle_val_i;
    }
    case SN_ENC_DICTIONARY: {
      int32_t idx = (uint16_t)rd_i16(d->body + ((int64_t)nnp << 1));
      if (d->dtype == SN_TYPE_INT32) return rd_i32(d->dict_entries + ((int64_t)idx << 2));
      return rd_i64(d->dict_entries + ((int64_t)idx << 3));
    }
    case SN_ENC_BIG_DICTIONARY: {
      int32_t idx = rd_i32(d->body + ((int64_t)nnp << 2));
      if (d->dtype == SN_TYPE_INT32) return rd_i32(d->dict_entries + ((int64_t)idx << 2));
      return rd_i64(d->dict_entries + ((int64_t)idx << 3));
    }
    case SN_ENC_BOOLEAN_BITSET:
      return bitset_get(d->body, nnp);
  }
  return 0;
}

static double dec_read_f64(sno_dec *d, int32_t nnp) {
  if (d->type_id == SN_ENC_UNCOMPRESSED) {
    if (d->dtype == SN_TYPE_DOUBLE) return rd_f64(d->body + ((int64_t)nnp << 3));
    if (d->dtype == SN_TYPE_FLOAT)  return rd_f32(d->body + ((int64_t)nnp << 2));
  }
  return (double)dec_read_i64(d, nnp);
}

/* dictionary index at nonNullPosition (DictionaryEncoding readDictionaryIndex);
 * null is NOT handled here (callers use null tracking; null index==dict_n
 * appears only via encoder-written index array for null rows) */
static inline int32_t dec_read_dict_index(sno_dec *d, int32_t nnp) {
  if (d->type_id == SN_ENC_DICTIONARY)
    return (uint16_t)rd_i16(d->body + ((int64_t)nnp << 1));
  return rd_i32(d->body + ((int64_t)nnp << 2));
}

/* string at nonNullPosition; only Uncompressed (sequential) and Dictionary */
static int dec_read_string(sno_dec *d, int32_t nnp, const uint8_t **s, int32_t *slen) {
  if (d->type_id == SN_ENC_DICTIONARY || d->type_id == SN_ENC_BIG_DICTIONARY) {
    int32_t idx = dec_read_dict_index(d, nnp);
    if (idx >= d->dict_n) { *s = NULL; *slen = 0; return SN_OK; } /* null slot */
    const uint8_t *e = d->dict_entries + d->dict_offsets[idx];
    *slen = rd_i32(e); *s = e + 4;
    return SN_OK;
  }
  if (d->type_id == SN_ENC_UNCOMPRESSED) {
    /* sequential cursor (Uncompressed.scala:117-160) */
    if (nnp != d->var_last_pos + 1) {
      if (nnp <= d->var_last_pos) return SN_ERR_BADARG; /* cursor cannot move back */
      while (d->var_last_pos + 1 < nnp) {
        int32_t sz = rd_i32(d->blob + d->var_cursor);
        d->var_cursor += 4 + sz;
        d->var_last_pos++;
      }
    }
    int32_t sz = rd_i32(d->blob + d->var_cursor);
    *s = d->blob + d->var_cursor + 4; *slen = sz;
    d->var_cursor += 4 + sz;
    d->var_last_pos = nnp;
    return SN_OK;
  }
  if (d->type_id == SN_ENC_RUNLENGTH) {
    if (d->rle_end_pos < nnp) {
      do {
        int32_t sz = rd_i32(d->blob + d->rle_cursor);
        d->rle_str = d->blob + d->rle_cursor + 4; d->rle_str_len = sz;
        d->rle_end_pos += rd_i32(d->blob + d->rle_cursor + 4 + sz);
        d->rle_cursor += 4 + sz + 4;
      } while (d->rle_end_pos < nnp);
    }
    *s = d->rle_str; *slen = d->rle_str_len;
    return SN_OK;
  }
  return SN_ERR_UNSUPPORTED;
}

/* =======================================================================
 * Delete-mask decoder (ColumnDeleteDecoder.scala:24-55)
 * ======================================================================= */
typedef struct {
  const uint8_t *pos;
  int32_t n, i;
  int32_t next_del;
} sno_del;

static int del_init(sno_del *del, const uint8_t *buf, int64_t len) {
  memset(del, 0, sizeof(*del));
  del->next_del = INT32_MAX;
  if (!buf) return SN_OK;
  if (len < 12) return SN_ERR_BADFORMAT;
  int32_t n = rd_i32(buf + 8);
  if (n < 0 || 12 + ((int64_t)n << 2) > len) return SN_ERR_BADFORMAT;
  del->pos = buf + 12; del->n = n; del->i = 0;
  del->next_del = n > 0 ? rd_i32(del->pos) : INT32_MAX;
  return SN_OK;
}
static inline int del_deleted(sno_del *del, int32_t ord) {
  if (del->next_del != ord) return 0;
  del->i++;
  del->next_del = del->i < del->n ? rd_i32(del->pos + ((int64_t)del->i << 2)) : INT32_MAX;
  return 1;
}

/* =======================================================================
 * Updated-column decoder: base + delta1/delta2 merge
 * (UpdatedColumnDecoder.scala:52-140, ColumnDeltaDecoder.scala:31-120)
 * ======================================================================= */
typedef struct {
  sno_dec dec;                /* the delta's own value decoder */
  const uint8_t *positions;   /* sorted batch ordinals */
  int32_t npos;
  int32_t pos_i;              /* next position index to read */
  int32_t decoder_position;   /* current delta entry (-1 before first) */
  int32_t non_null_position;  /* current non-null delta entry */
  int     not_null;
} sno_delta;

static int delta_init(sno_delta *dd, int32_t dtype, const uint8_t *buf, int64_t len) {
  memset(dd, 0, sizeof(*dd));
  dd->decoder_position = -1;
  dd->non_null_position = -1;
  return dec_init(&dd->dec, dtype, buf, len, 1, &dd->positions, &dd->npos);
}
static inline int32_t delta_read_position(sno_delta *dd) {
  return dd->pos_i < dd->npos ? rd_i32(dd->positions + ((int64_t)dd->pos_i << 2)) : INT32_MAX;
}
static inline void delta_move_cursor(sno_delta *dd) {
  dd->pos_i++;
  dd->decoder_position++;
  dd->not_null = !dec_is_null_at(&dd->dec, dd->decoder_position);
  if (dd->not_null) dd->non_null_position++;
}

typedef struct {
  sno_delta d1, d2;
  int has1, has2;
  int32_t next1, next2;
  sno_delta *current;
  int32_t next_updated;      /* next batch ordinal with an update */
} sno_upd;

/* moveToNextUpdatedPosition (UpdatedColumnDecoder.scala:69-96) */
static int32_t upd_move_next(sno_upd *u) {
  int32_t next = INT32_MAX;
  int first = 0;
  if (u->has1 && u->next1 != INT32_MAX) { next = u->next1; first = 1; }
  if (u->has2) {
    if (u->next2 <= next) {
      if (u->next2 < next) { next = u->next2; u->current = &u->d2; first = 0; }
      /* skip on equality (delta1 wins) */
      delta_move_cursor(&u->d2);
      u->next2 = delta_read_position(&u->d2);
    }
  }
  if (first) {
    u->current = &u->d1;
    delta_move_cursor(&u->d1);
    u->next1 = delta_read_position(&u->d1);
  }
  return next;
}

static int upd_init(sno_upd *u, int32_t dtype, const uint8_t *b1, int64_t l1,
                    const uint8_t *b2, int64_t l2) {
  memset(u, 0, sizeof(*u));
  u->next1 = u->next2 = INT32_MAX;
  int rc;
  if (b1) { u->has1 = 1; if ((rc = delta_init(&u->d1, dtype, b1, l1)) != SN_OK) return rc;
            u->next1 = delta_read_position(&u->d1); }
  if (b2) { u->has2 = 1; if ((rc = delta_init(&u->d2, dtype, b2, l2)) != SN_OK) return rc;
            u->next2 = delta_read_position(&u->d2); }
  u->next_updated = upd_move_next(u);
  return SN_OK;
}
static void upd_free(sno_upd *u) {
  if (u->has1) dec_free(&u->d1.dec);
  if (u->has2) dec_free(&u->d2.dec);
}
/* returns: 0 = use base value, 1 = updated (fills *is_null and value reads
 * pending on current delta), for ordinal `ord` visited in ascending order */
static inline int upd_at(sno_upd *u, int32_t ord) {
  if (u->next_updated != ord) return 0;
  return 1;
}
static inline void upd_advance(sno_upd *u) { u->next_updated = upd_move_next(u); }

/* =======================================================================
 * Batch container
 * ======================================================================= */
typedef struct {
  int32_t num_rows;
  int     has_deltas;
  uint8_t **cols;        /* owned copies */
  int64_t *col_lens;
  uint8_t *del;   int64_t del_len;
  uint8_t **deltas;      /* [ncols*2] */
  int64_t *delta_lens;
  uint8_t *stats; int64_t stats_len;
} sno_batch;

typedef struct sno_table {
  int32_t ncols;
  int32_t *dtypes;
  uint8_t *nullable;
  int32_t nbatches, cap;
  sno_batch *batches;
  /* broadcast dimension (HashJoinExec stand-in): open-address key table +
   * per-key attribute strings (independent restatement for parity checks) */
  int64_t *dim_hk;         /* capacity dim_cap, sentinel INT64_MIN */
  int32_t *dim_hidx;       /* key ordinal per slot */
  int64_t  dim_cap;
  char    *dim_attr_pay;
  int32_t *dim_attr_off;   /* nkeys+1 offsets into dim_attr_pay */
  int64_t  dim_n;
} sno_table;

SNO_EXPORT sno_table *sno_table_create(int32_t ncols, const int32_t *dtypes,
                                       const uint8_t *nullable) {
  if (ncols <= 0 || ncols > SN_MAX_PREDS + 64) return NULL;
  sno_table *t = (sno_table *)calloc(1, sizeof(sno_table));
  t->ncols = ncols;
  t->dtypes = (int32_t *)malloc(sizeof(int32_t) * (size_t)ncols);
  t->nullable = (uint8_t *)malloc((size_t)ncols);
  memcpy(t->dtypes, dtypes, sizeof(int32_t) * (size_t)ncols);
  if (nullable) memcpy(t->nullable, nullable, (size_t)ncols);
  else memset(t->nullable, 0, (size_t)ncols);
  return t;
}

static uint8_t *dup_buf(const void *p, int64_t n) {
  if (!p || n <= 0) return NULL;
  uint8_t *b = (uint8_t *)malloc((size_t)n);
  memcpy(b, p, (size_t)n);
  return b;
}

/* LZ4 wrapper (CompressionUtils.scala:53-61): [-codecId][ulen][payload] */
typedef int (*sno_lz4_fn)(const char *, char *, int, int);
static sno_lz4_fn sno_get_lz4(void) {
  static sno_lz4_fn fn = NULL;
  static int tried = 0;
  if (!tried) {
    tried = 1;
    void *h = dlopen("liblz4.so.1", RTLD_NOW);
    if (!h) h = dlopen("liblz4.so", RTLD_NOW);
    if (h) fn = (sno_lz4_fn)dlsym(h, "LZ4_decompress_safe");
  }
  return fn;
}

/* Raw Snappy block decode (google/snappy format_description.txt — the raw
 * block format snappy-java emits; reference codec 2,
 * CompressionCodecId.scala:31).  varint32 uncompressed length, then
 * literal/copy elements; copies may overlap (byte-by-byte semantics). */
static int sno_snappy_decompress(const uint8_t *src, int64_t slen,
                                 uint8_t *dst, uint32_t ulen) {
  int64_t ip = 0;
  uint32_t declared = 0;
  int shift = 0;
  while (ip < slen) {
    uint8_t b = src[ip++];
    declared |= (uint32_t)(b & 0x7fu) << shift;
    if (!(b & 0x80u)) break;
    shift += 7;
    if (shift > 28) return -1;
  }
  if (declared != ulen) return -1;
  uint32_t op = 0;
  while (ip < slen) {
    uint8_t tag = src[ip++];
    if ((tag & 3) == 0) {
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
    } else {
      uint32_t n, off;
      if ((tag & 3) == 1) {
        if (ip >= slen) return -1;
        n = ((uint32_t)(tag >> 2) & 7u) + 4;
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

/* returns malloc'd decompressed buffer (caller frees) or NULL if plain */
static uint8_t *sno_maybe_decompress(const uint8_t *blob, int64_t len,
                                     int64_t *out_len, int *err) {
  *err = 0;
  if (len < 8 || rd_i32(blob) >= 0) return NULL;
  int32_t codec = -rd_i32(blob);
  int32_t ulen = rd_i32(blob + 4);
  if ((codec != 1 && codec != 2) || ulen <= 0) {
    *err = SN_ERR_UNSUPPORTED;
    return NULL;
  }
  uint8_t *buf = (uint8_t *)malloc((size_t)ulen);
  if (codec == 1) {
    sno_lz4_fn fn = sno_get_lz4();
    if (!fn) { free(buf); *err = SN_ERR_UNSUPPORTED; return NULL; }
    int n = fn((const char *)blob + 8, (char *)buf, (int)(len - 8), ulen);
    if (n != ulen) { free(buf); *err = SN_ERR_BADFORMAT; return NULL; }
  } else {
    if (sno_snappy_decompress(blob + 8, len - 8, buf, (uint32_t)ulen) != 0) {
      free(buf); *err = SN_ERR_BADFORMAT; return NULL;
    }
  }
  *out_len = ulen;
  return buf;
}

SNO_EXPORT int32_t sno_table_add_batch(sno_table *t, int32_t num_rows,
    const sn_buf *cols, const sn_buf *stats, const sn_buf *delete_mask,
    const sn_buf *deltas /* [ncols*2] or NULL */) {
  if (!t || num_rows == 0 || !cols) return SN_ERR_BADARG;
  if (t->nbatches == t->cap) {
    t->cap = t->cap ? t->cap * 2 : 16;
    t->batches = (sno_batch *)realloc(t->batches, sizeof(sno_batch) * (size_t)t->cap);
  }
  sno_batch *b = &t->batches[t->nbatches];
  memset(b, 0, sizeof(*b));
  /* negative batchCount convention => has update deltas (ColumnTableScan.scala:524-528) */
  b->has_deltas = num_rows < 0;
  b->num_rows = num_rows < 0 ? -num_rows : num_rows;
  b->cols = (uint8_t **)calloc((size_t)t->ncols, sizeof(void *));
  b->col_lens = (int64_t *)calloc((size_t)t->ncols, sizeof(int64_t));
  for (int i = 0; i < t->ncols; i++) {
    if (!cols[i].data || cols[i].len < 8) { return SN_ERR_BADARG; }
    int err = 0;
    int64_t dlen = 0;
    uint8_t *dec = sno_maybe_decompress((const uint8_t *)cols[i].data,
                                        cols[i].len, &dlen, &err);
    if (err) return err;
    if (dec) { b->cols[i] = dec; b->col_lens[i] = dlen; }
    else {
      b->cols[i] = dup_buf(cols[i].data, cols[i].len);
      b->col_lens[i] = cols[i].len;
    }
  }
  if (stats && stats->data) { b->stats = dup_buf(stats->data, stats->len); b->stats_len = stats->len; }
  if (delete_mask && delete_mask->data) {
    b->del = dup_buf(delete_mask->data, delete_mask->len); b->del_len = delete_mask->len;
  }
  if (deltas) {
    b->deltas = (uint8_t **)calloc((size_t)t->ncols * 2, sizeof(void *));
    b->delta_lens = (int64_t *)calloc((size_t)t->ncols * 2, sizeof(int64_t));
    for (int i = 0; i < t->ncols * 2; i++) {
      if (deltas[i].data) {
        b->deltas[i] = dup_buf(deltas[i].data, deltas[i].len);
        b->delta_lens[i] = deltas[i].len;
      }
    }
  }
  t->nbatches++;
  return SN_OK;
}

static uint64_t sno_mix64(uint64_t x) {
  x += 0x9E3779B97f4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

SNO_EXPORT int32_t sno_table_set_dim(sno_table *t, const int64_t *keys,
                                     int64_t n, const char *attr_payload,
                                     const int32_t *attr_lens) {
  if (!t || !keys || n <= 0) return SN_ERR_BADARG;
  int64_t cap = 2;
  while (cap < 2 * n + 1) cap <<= 1;
  t->dim_hk = (int64_t *)malloc((size_t)cap * 8);
  t->dim_hidx = (int32_t *)malloc((size_t)cap * 4);
  for (int64_t i = 0; i < cap; i++) t->dim_hk[i] = INT64_MIN;
  t->dim_cap = cap;
  t->dim_n = n;
  int64_t total = 0;
  if (attr_payload && attr_lens)
    for (int64_t i = 0; i < n; i++) total += attr_lens[i];
  t->dim_attr_pay = (char *)malloc((size_t)(total ? total : 1));
  t->dim_attr_off = (int32_t *)malloc((size_t)(n + 1) * 4);
  int64_t off = 0;
  for (int64_t i = 0; i < n; i++) {
    t->dim_attr_off[i] = (int32_t)off;
    if (attr_payload && attr_lens) {
      memcpy(t->dim_attr_pay + off, attr_payload + off, (size_t)attr_lens[i]);
      off += attr_lens[i];
    }
    uint64_t h = sno_mix64((uint64_t)keys[i]) & (uint64_t)(cap - 1);
    while (t->dim_hk[h] != INT64_MIN) {
      if (t->dim_hk[h] == keys[i]) return SN_ERR_BADARG;
      h = (h + 1) & (uint64_t)(cap - 1);
    }
    t->dim_hk[h] = keys[i];
    t->dim_hidx[h] = (int32_t)i;
  }
  t->dim_attr_off[n] = (int32_t)off;
  return SN_OK;
}

static int dim_lookup(const sno_table *t, int64_t key) {
  if (!t->dim_hk) return -1;
  uint64_t h = sno_mix64((uint64_t)key) & (uint64_t)(t->dim_cap - 1);
  while (1) {
    int64_t k0 = t->dim_hk[h];
    if (k0 == key) return t->dim_hidx[h];
    if (k0 == INT64_MIN) return -1;
    h = (h + 1) & (uint64_t)(t->dim_cap - 1);
  }
}

SNO_EXPORT void sno_table_destroy(sno_table *t) {
  if (!t) return;
  for (int32_t bi = 0; bi < t->nbatches; bi++) {
    sno_batch *b = &t->batches[bi];
    for (int i = 0; i < t->ncols; i++) free(b->cols[i]);
    free(b->cols); free(b->col_lens);
    free(b->del); free(b->stats);
    if (b->deltas) {
      for (int i = 0; i < t->ncols * 2; i++) free(b->deltas[i]);
      free(b->deltas); free(b->delta_lens);
    }
  }
  free(t->batches); free(t->dtypes); free(t->nullable);
  free(t->dim_hk); free(t->dim_hidx); free(t->dim_attr_pay); free(t->dim_attr_off);
  free(t);
}

/* =======================================================================
 * Stats row (UnsafeRow) parse + batch skip
 * ======================================================================= */
typedef struct {
  int32_t batch_count;      /* signed as stored */
  int     valid;
  /* per column */
  double  *lower_d, *upper_d;
  int64_t *lower_i, *upper_i;
  int32_t *null_count;
  uint8_t *bounds_null;     /* 1 = lower/upper is SQL NULL */
} sno_stats;

static int stats_parse(const uint8_t *blob, int64_t len, int32_t ncols,
                       const int32_t *dtypes, sno_stats *st) {
  memset(st, 0, sizeof(*st));
  if (!blob || len < 8) return SN_ERR_BADARG;
  int32_t num_fields = ncols * 3 + 1;
  int32_t null_words = (num_fields + 63) >> 6;
  int64_t fixed = (int64_t)null_words * 8 + (int64_t)num_fields * 8;
  if (len < fixed) return SN_ERR_BADFORMAT;
  const uint8_t *bits = blob;
  const uint8_t *slots = blob + (int64_t)null_words * 8;
  st->lower_d = (double *)calloc((size_t)ncols, 8);
  st->upper_d = (double *)calloc((size_t)ncols, 8);
  st->lower_i = (int64_t *)calloc((size_t)ncols, 8);
  st->upper_i = (int64_t *)calloc((size_t)ncols, 8);
  st->null_count = (int32_t *)calloc((size_t)ncols, 4);
  st->bounds_null = (uint8_t *)malloc((size_t)ncols);
  st->batch_count = rd_i32(slots);   /* field 0, int in 8B slot */
  for (int c = 0; c < ncols; c++) {
    int f_lo = 1 + c * 3, f_hi = 2 + c * 3, f_nc = 3 + c * 3;
    int lo_null = bitset_get(bits, f_lo), hi_null = bitset_get(bits, f_hi);
    st->bounds_null[c] = (uint8_t)(lo_null || hi_null);
    st->null_count[c] = bitset_get(bits, f_nc) ? 0 : rd_i32(slots + (int64_t)f_nc * 8);
    if (!st->bounds_null[c]) {
      switch (dtypes[c]) {
        case SN_TYPE_DOUBLE:
          st->lower_d[c] = rd_f64(slots + (int64_t)f_lo * 8);
          st->upper_d[c] = rd_f64(slots + (int64_t)f_hi * 8);
          break;
        case SN_TYPE_FLOAT:
          st->lower_d[c] = rd_f32(slots + (int64_t)f_lo * 8);
          st->upper_d[c] = rd_f32(slots + (int64_t)f_hi * 8);
          break;
        case SN_TYPE_INT32: case SN_TYPE_INT16: case SN_TYPE_INT8: case SN_TYPE_BOOL:
          st->lower_i[c] = rd_i32(slots + (int64_t)f_lo * 8);
          st->upper_i[c] = rd_i32(slots + (int64_t)f_hi * 8);
          break;
        case SN_TYPE_INT64:
          st->lower_i[c] = rd_i64(slots + (int64_t)f_lo * 8);
          st->upper_i[c] = rd_i64(slots + (int64_t)f_hi * 8);
          break;
        default:
          st->bounds_null[c] = 1; /* string bounds unused by round-1 plans */
      }
    }
  }
  st->valid = 1;
  return SN_OK;
}
static void stats_free(sno_stats *st) {
  free(st->lower_d); free(st->upper_d); free(st->lower_i); free(st->upper_i);
  free(st->null_count); free(st->bounds_null);
}

/* conservative batch-skip: true => NO row can satisfy the conjunction
 * (generateStatPredicate semantics, ColumnTableScan.scala:845-891) */
static int stats_skip(const sno_stats *st, const sn_plan *p, const int32_t *dtypes,
                      int batch_rows) {
  if (!st->valid) return 0;
  for (int i = 0; i < p->npreds; i++) {
    const sn_pred *pr = &p->preds[i];
    int c = pr->col;
    if (st->null_count[c] >= batch_rows && batch_rows > 0) return 1; /* all NULL */
    if (st->bounds_null[c]) continue;
    int is_d = (dtypes[c] == SN_TYPE_DOUBLE || dtypes[c] == SN_TYPE_FLOAT);
    if (is_d) {
      if (pr->has_lo && (pr->lo_strict ? st->upper_d[c] <= pr->lo_d
                                       : st->upper_d[c] <  pr->lo_d)) return 1;
      if (pr->has_hi && (pr->hi_strict ? st->lower_d[c] >= pr->hi_d
                                       : st->lower_d[c] >  pr->hi_d)) return 1;
    } else {
      if (pr->has_lo && (pr->lo_strict ? st->upper_i[c] <= pr->lo_i
                                       : st->upper_i[c] <  pr->lo_i)) return 1;
      if (pr->has_hi && (pr->hi_strict ? st->lower_i[c] >= pr->hi_i
                                       : st->lower_i[c] >  pr->hi_i)) return 1;
    }
  }
  return 0;
}

/* =======================================================================
 * Group table (small: linear/hash over interned key pairs)
 * ======================================================================= */
typedef struct {
  char keys[SN_MAX_GROUPS][SN_KEY_MAX];
  uint8_t key_null[SN_MAX_GROUPS];
  double sums[SN_MAX_AGGS];
  double counts[SN_MAX_AGGS]; /* non-null input counts (for AVG/SUM null) */
  double rowcount;
} sno_group;

typedef struct {
  sno_group *groups;
  int32_t n, cap;
  int32_t ngroup_cols, naggs;
  int32_t max_groups;     /* hard cardinality bound */
  /* hash index: key -> group slot */
  int32_t *index;  /* size index_cap, -1 empty */
  int32_t index_cap;
} sno_gtab;

static uint64_t key_hash(const char (*keys)[SN_KEY_MAX], const uint8_t *knull, int n) {
  uint64_t h = 1469598103934665603ULL;
  for (int i = 0; i < n; i++) {
    h ^= knull[i]; h *= 1099511628211ULL;
    for (const char *s = keys[i]; *s; s++) { h ^= (uint8_t)*s; h *= 1099511628211ULL; }
    h ^= 0xff; h *= 1099511628211ULL;
  }
  return h;
}

/* max_groups 0 -> default 1<<22: the oracle follows ByteBufferHashMap's
 * grow-and-rehash (ByteBufferHashMap.scala:245-292) so cardinalities well
 * beyond the engine's one-page result bound are checkable */
static int gtab_init(sno_gtab *g, int ngroup_cols, int naggs) {
  memset(g, 0, sizeof(*g));
  g->ngroup_cols = ngroup_cols; g->naggs = naggs;
  g->max_groups = 1 << 22;
  g->cap = 64;
  g->groups = (sno_group *)calloc((size_t)g->cap, sizeof(sno_group));
  g->index_cap = 4 * SN_MAX_GROUP_SLOTS;
  g->index = (int32_t *)malloc(sizeof(int32_t) * (size_t)g->index_cap);
  memset(g->index, 0xff, sizeof(int32_t) * (size_t)g->index_cap);
  return SN_OK;
}
static void gtab_free(sno_gtab *g) { free(g->groups); free(g->index); }

static void gtab_rehash(sno_gtab *g) {
  int32_t ncap = g->index_cap * 2;
  int32_t *ni = (int32_t *)malloc(sizeof(int32_t) * (size_t)ncap);
  if (!ni) return;                 /* keep probing the old (denser) index */
  memset(ni, 0xff, sizeof(int32_t) * (size_t)ncap);
  for (int32_t gi = 0; gi < g->n; gi++) {
    uint64_t h = key_hash((const char (*)[SN_KEY_MAX])g->groups[gi].keys,
                          g->groups[gi].key_null, g->ngroup_cols);
    int32_t s = (int32_t)(h & (uint64_t)(ncap - 1));
    while (ni[s] >= 0) s = (s + 1) & (ncap - 1);
    ni[s] = gi;
  }
  free(g->index);
  g->index = ni;
  g->index_cap = ncap;
}

static sno_group *gtab_get(sno_gtab *g, const char (*keys)[SN_KEY_MAX],
                           const uint8_t *knull) {
  if (g->n * 2 >= g->index_cap) gtab_rehash(g);
  uint64_t h = key_hash(keys, knull, g->ngroup_cols);
  int32_t slot = (int32_t)(h & (uint64_t)(g->index_cap - 1));
  while (1) {
    int32_t gi = g->index[slot];
    if (gi < 0) {
      if (g->n >= g->max_groups) return NULL;
      if (g->n == g->cap) {
        g->cap *= 2;
        g->groups = (sno_group *)realloc(g->groups, sizeof(sno_group) * (size_t)g->cap);
        memset(g->groups + g->n, 0, sizeof(sno_group) * (size_t)(g->cap - g->n));
      }
      sno_group *grp = &g->groups[g->n];
      memset(grp, 0, sizeof(*grp));
      for (int i = 0; i < g->ngroup_cols; i++) {
        strncpy(grp->keys[i], keys[i], SN_KEY_MAX - 1);
        grp->key_null[i] = knull[i];
      }
      g->index[slot] = g->n;
      return &g->groups[g->n++];
    }
    sno_group *grp = &g->groups[gi];
    int match = 1;
    for (int i = 0; i < g->ngroup_cols && match; i++) {
      if (grp->key_null[i] != knull[i]) match = 0;
      else if (!knull[i] && strncmp(grp->keys[i], keys[i], SN_KEY_MAX) != 0) match = 0;
    }
    if (match) return grp;
    slot = (slot + 1) & (g->index_cap - 1);
  }
}

/* =======================================================================
 * The evaluator — restates the generated per-partition loop
 * (ColumnTableScan.scala:636-815 + SnappyHashAggregateExec accumulate)
 * ======================================================================= */
typedef struct {
  int needed;               /* referenced by plan */
  sno_dec dec;
  sno_upd upd;
  int has_upd;
  int is_group;             /* group-by column */
} col_state;

static int eval_batch(const sno_table *t, const sno_batch *b, const sn_plan *p,
                      sno_gtab *g, int64_t *rows_scanned, int64_t *rows_passed) {
  int rc = SN_OK;
  int nc = t->ncols;
  col_state *cs = (col_state *)calloc((size_t)nc, sizeof(col_state));
  const int has_join = p->join_dim != SN_JOIN_NONE && t->dim_hk != NULL;
  const int join_group = has_join && p->join_mode == SN_JOIN_GROUP;
  if (has_join) cs[p->join_fact_col].needed = 1;
  for (int i = 0; i < p->npreds; i++) cs[p->preds[i].col].needed = 1;
  for (int i = 0; i < p->ngroup; i++) { cs[p->group_cols[i]].needed = 1; cs[p->group_cols[i]].is_group = 1; }
  for (int a = 0; a < p->naggs; a++)
    for (int f = 0; f < p->aggs[a].nfactors; f++) cs[p->aggs[a].factors[f].col].needed = 1;

  sno_del del;
  if ((rc = del_init(&del, b->del, b->del_len)) != SN_OK) goto done;

  for (int c = 0; c < nc; c++) {
    if (!cs[c].needed) continue;
    if ((rc = dec_init(&cs[c].dec, t->dtypes[c], b->cols[c], b->col_lens[c], 0, NULL, NULL)) != SN_OK)
      goto done;
    if (b->deltas && (b->deltas[c * 2] || b->deltas[c * 2 + 1])) {
      cs[c].has_upd = 1;
      if ((rc = upd_init(&cs[c].upd, t->dtypes[c],
                         b->deltas[c * 2], b->delta_lens[c * 2],
                         b->deltas[c * 2 + 1], b->delta_lens[c * 2 + 1])) != SN_OK)
        goto done;
    }
  }

  for (int32_t ord = 0; ord < b->num_rows; ord++) {
    /* per-row: advance every needed column's sequential state exactly once */
    int deleted = del_deleted(&del, ord);
    (*rows_scanned) += !deleted;

    /* value snapshot per needed column */
    double  val_d[64]; int64_t val_i[64]; uint8_t val_null[64];
    const uint8_t *val_s[64]; int32_t val_slen[64];
    int row_ok = 1;

    for (int c = 0; c < nc; c++) {
      if (!cs[c].needed) continue;
      sno_dec *d = &cs[c].dec;
      int base_null = dec_null_advance(d, ord);    /* must advance even when deleted/filtered */
      /* nonNullPosition = ordinal minus nulls strictly before it
       * (NullableDecoder numNulls semantics, ColumnEncoding.scala:1110-1135) */
      int32_t nnp = ord - d->nulls_before;
      int use_delta = 0;
      if (cs[c].has_upd && upd_at(&cs[c].upd, ord)) use_delta = 1;

      if (deleted) { if (use_delta) upd_advance(&cs[c].upd); continue; }

      if (use_delta) {
        sno_delta *dd = cs[c].upd.current;
        val_null[c] = (uint8_t)!dd->not_null;
        if (dd->not_null) {
          int32_t dp = dd->non_null_position;
          if (t->dtypes[c] == SN_TYPE_DOUBLE || t->dtypes[c] == SN_TYPE_FLOAT) {
            val_d[c] = dec_read_f64(&dd->dec, dp); val_i[c] = (int64_t)val_d[c];
          } else if (t->dtypes[c] == SN_TYPE_STRING) {
            if ((rc = dec_read_string(&dd->dec, dp, &val_s[c], &val_slen[c])) != SN_OK) goto done;
          } else {
            val_i[c] = dec_read_i64(&dd->dec, dp); val_d[c] = (double)val_i[c];
          }
        }
        upd_advance(&cs[c].upd);
      } else {
        val_null[c] = (uint8_t)base_null;
        if (!base_null) {
          if (t->dtypes[c] == SN_TYPE_DOUBLE || t->dtypes[c] == SN_TYPE_FLOAT) {
            val_d[c] = dec_read_f64(d, nnp); val_i[c] = (int64_t)val_d[c];
          } else if (t->dtypes[c] == SN_TYPE_STRING) {
            if ((rc = dec_read_string(d, nnp, &val_s[c], &val_slen[c])) != SN_OK) goto done;
            if (val_s[c] == NULL) val_null[c] = 1;  /* dict null slot */
          } else {
            val_i[c] = dec_read_i64(d, nnp); val_d[c] = (double)val_i[c];
          }
        }
      }
    }
    if (deleted) continue;

    /* predicates (NULL comparison => false) */
    for (int i = 0; i < p->npreds && row_ok; i++) {
      const sn_pred *pr = &p->preds[i];
      int c = pr->col;
      if (val_null[c]) { row_ok = 0; break; }
      if (pr->in_n > 0) {
        /* IN-list membership (Q12/Q19-class) */
        int hit = 0;
        if (t->dtypes[c] == SN_TYPE_STRING) {
          for (int32_t ii = 0; ii < pr->in_n && !hit; ii++)
            if (pr->in_s && pr->in_s_len &&
                val_slen[c] == pr->in_s_len[ii] &&
                memcmp(val_s[c], pr->in_s[ii], (size_t)val_slen[c]) == 0)
              hit = 1;
        } else {
          for (int32_t ii = 0; ii < pr->in_n && !hit; ii++)
            if (pr->in_i && pr->in_i[ii] == val_i[c]) hit = 1;
        }
        if (!hit) row_ok = 0;
        continue;
      }
      if (pr->str_eq && pr->str_len > 0) {
        /* dictionary string equality (engine pushes this down to a dict-id
         * compare; the oracle compares the decoded bytes) */
        if (t->dtypes[c] != SN_TYPE_STRING || val_slen[c] != pr->str_len ||
            memcmp(val_s[c], pr->str_eq, (size_t)pr->str_len) != 0)
          row_ok = 0;
        continue;
      }
      int is_d = (t->dtypes[c] == SN_TYPE_DOUBLE || t->dtypes[c] == SN_TYPE_FLOAT);
      if (is_d) {
        double v = val_d[c];
        if (pr->has_lo && (pr->lo_strict ? !(v > pr->lo_d) : !(v >= pr->lo_d))) row_ok = 0;
        if (pr->has_hi && (pr->hi_strict ? !(v < pr->hi_d) : !(v <= pr->hi_d))) row_ok = 0;
      } else {
        int64_t v = val_i[c];
        if (pr->has_lo && (pr->lo_strict ? !(v > pr->lo_i) : !(v >= pr->lo_i))) row_ok = 0;
        if (pr->has_hi && (pr->hi_strict ? !(v < pr->hi_i) : !(v <= pr->hi_i))) row_ok = 0;
      }
    }
    /* broadcast-dimension probe (inner join on unique key) */
    int dim_ord = -1;
    if (row_ok && has_join) {
      if (val_null[p->join_fact_col]) row_ok = 0;
      else {
        dim_ord = dim_lookup(t, val_i[p->join_fact_col]);
        if (dim_ord < 0) row_ok = 0;
      }
    }
    if (!row_ok) continue;
    (*rows_passed)++;

    /* group lookup */
    char keys[SN_MAX_GROUPS][SN_KEY_MAX];
    uint8_t knull[SN_MAX_GROUPS];
    memset(knull, 0, sizeof(knull));
    int kbase = 0;
    if (join_group) {
      int32_t a0 = t->dim_attr_off[dim_ord], a1 = t->dim_attr_off[dim_ord + 1];
      int32_t L = a1 - a0 < SN_KEY_MAX - 1 ? a1 - a0 : SN_KEY_MAX - 1;
      memcpy(keys[0], t->dim_attr_pay + a0, (size_t)L);
      keys[0][L] = 0;
      kbase = 1;   /* composite GROUP BY dim_attr, fact_col: attr first */
    }
    for (int i = 0; i < p->ngroup && kbase + i < SN_MAX_GROUPS; i++) {
      int c = p->group_cols[i];
      int ki = kbase + i;
      if (val_null[c]) { knull[ki] = 1; keys[ki][0] = 0; }
      else if (t->dtypes[c] != SN_TYPE_STRING) {
        /* integer group key: decimal text (SHAMapAccessor serializes the
         * raw fixed-width key; text keeps the partial-block format shared) */
        snprintf(keys[ki], SN_KEY_MAX, "%lld", (long long)val_i[c]);
      } else {
        int32_t L = val_slen[c] < SN_KEY_MAX - 1 ? val_slen[c] : SN_KEY_MAX - 1;
        memcpy(keys[ki], val_s[c], (size_t)L); keys[ki][L] = 0;
      }
    }
    sno_group *grp = gtab_get(g, keys, knull);
    if (!grp) { rc = SN_ERR_OVERFLOW; goto done; }
    grp->rowcount += 1.0;

    /* aggregates */
    for (int a = 0; a < p->naggs; a++) {
      const sn_agg *ag = &p->aggs[a];
      if (ag->kind == SN_AGG_COUNT_STAR) { grp->sums[a] += 1.0; grp->counts[a] += 1.0; continue; }
      int anynull = 0;
      double v = 1.0;
      for (int f = 0; f < ag->nfactors; f++) {
        int c = ag->factors[f].col;
        if (val_null[c]) { anynull = 1; break; }
        v *= ag->factors[f].add + ag->factors[f].mul * val_d[c];
      }
      if (!anynull) {
        if (ag->kind == SN_AGG_MIN)
          grp->sums[a] = grp->counts[a] > 0 ? fmin(grp->sums[a], v) : v;
        else if (ag->kind == SN_AGG_MAX)
          grp->sums[a] = grp->counts[a] > 0 ? fmax(grp->sums[a], v) : v;
        else
          grp->sums[a] += v;
        grp->counts[a] += 1.0;
      }
    }
  }

done:
  for (int c = 0; c < nc; c++) {
    if (cs[c].needed) dec_free(&cs[c].dec);
    if (cs[c].has_upd) upd_free(&cs[c].upd);
  }
  free(cs);
  return rc;
}

/* -------- fast path: clean batches (no nulls/deltas/deletes), numeric
 * uncompressed predicate/aggregate columns, dict16 group columns.  This is
 * the CPU-baseline leg's hot loop — a tight restatement of the generated
 * WholeStageCodegen loop shape so the reported baseline is honest (the
 * generic path above pays per-row dispatch the JVM's generated code does
 * not). -------- */
typedef struct { const uint8_t *base; int w; int is_f; } fcol;

static int eval_batch_fast(const sno_table *t, const sno_batch *b,
                           const sn_plan *p, sno_gtab *g,
                           int64_t *rows_scanned, int64_t *rows_passed) {
  if (b->has_deltas || b->del) return 0;
  if (p->npreds > SN_MAX_PREDS || p->naggs > SN_MAX_AGGS) return 0;
  for (int a = 0; a < p->naggs; a++)   /* MIN/MAX keep the generic path */
    if (p->aggs[a].kind == SN_AGG_MIN || p->aggs[a].kind == SN_AGG_MAX) return 0;
  for (int i = 0; i < p->npreds; i++)  /* IN-lists keep the generic path */
    if (p->preds[i].in_n > 0) return 0;
  const int has_join = p->join_dim != SN_JOIN_NONE && t->dim_hk != NULL;
  if (has_join) return 0;   /* join keeps the generic path */

  /* resolve typed column views */
  fcol cols[64];
  const uint8_t *gidx[SN_MAX_GROUPS];
  int gdictn[SN_MAX_GROUPS];
  const uint8_t *gdict_entries[SN_MAX_GROUPS];
  for (int c = 0; c < t->ncols; c++) cols[c].base = NULL;

  for (int i = 0; i < p->npreds; i++) {
    int c = p->preds[i].col;
    sno_dec d;
    if (dec_init(&d, t->dtypes[c], b->cols[c], b->col_lens[c], 0, NULL, NULL) != SN_OK)
      return 0;
    int bad = d.type_id != SN_ENC_UNCOMPRESSED || d.null_words != NULL ||
              t->dtypes[c] == SN_TYPE_STRING;
    cols[c].base = d.body;
    cols[c].w = type_width(t->dtypes[c]);
    cols[c].is_f = t->dtypes[c] == SN_TYPE_DOUBLE || t->dtypes[c] == SN_TYPE_FLOAT;
    dec_free(&d);
    if (bad) return 0;
  }
  for (int a = 0; a < p->naggs; a++)
    for (int f = 0; f < p->aggs[a].nfactors; f++) {
      int c = p->aggs[a].factors[f].col;
      sno_dec d;
      if (dec_init(&d, t->dtypes[c], b->cols[c], b->col_lens[c], 0, NULL, NULL) != SN_OK)
        return 0;
      int bad = d.type_id != SN_ENC_UNCOMPRESSED || d.null_words != NULL ||
                t->dtypes[c] == SN_TYPE_STRING;
      cols[c].base = d.body;
      cols[c].w = type_width(t->dtypes[c]);
      cols[c].is_f = cols[c].is_f || t->dtypes[c] == SN_TYPE_DOUBLE ||
                     t->dtypes[c] == SN_TYPE_FLOAT;
      cols[c].is_f = (t->dtypes[c] == SN_TYPE_DOUBLE || t->dtypes[c] == SN_TYPE_FLOAT);
      dec_free(&d);
      if (bad) return 0;
    }
  int nslots = 1;
  for (int i = 0; i < p->ngroup; i++) {
    int c = p->group_cols[i];
    if (t->dtypes[c] != SN_TYPE_STRING) return 0;
    sno_dec d;
    if (dec_init(&d, t->dtypes[c], b->cols[c], b->col_lens[c], 0, NULL, NULL) != SN_OK)
      return 0;
    int bad = d.type_id != SN_ENC_DICTIONARY || d.null_words != NULL ||
              d.dict_n > 64;
    gidx[i] = d.body;
    gdictn[i] = d.dict_n;
    gdict_entries[i] = d.dict_entries;
    /* keep dict offsets alive: re-derive below from entries (lens inline) */
    dec_free(&d);
    if (bad) return 0;
    nslots *= gdictn[i];
  }
  if (nslots > 4096) return 0;

  /* canonical predicate/aggregate forms (one-time per batch) */
  double plo[SN_MAX_PREDS], phi[SN_MAX_PREDS];
  const uint8_t *pbase[SN_MAX_PREDS];
  int pw[SN_MAX_PREDS], pisf[SN_MAX_PREDS];
  for (int i = 0; i < p->npreds; i++) {
    const sn_pred *pr = &p->preds[i];
    int c = pr->col;
    pbase[i] = cols[c].base; pw[i] = cols[c].w; pisf[i] = cols[c].is_f;
    double lo = pr->has_lo ? (pisf[i] ? pr->lo_d : (double)pr->lo_i) : -INFINITY;
    double hi = pr->has_hi ? (pisf[i] ? pr->hi_d : (double)pr->hi_i) : INFINITY;
    if (pr->has_lo && pr->lo_strict) lo = nextafter(lo, INFINITY);
    if (pr->has_hi && pr->hi_strict) hi = nextafter(hi, -INFINITY);
    if (!pisf[i] && t->dtypes[c] == SN_TYPE_INT64) return 0; /* exactness */
    plo[i] = lo; phi[i] = hi;
  }
  double aa[SN_MAX_AGGS][3], am[SN_MAX_AGGS][3];
  const uint8_t *ab[SN_MAX_AGGS][3];
  int aw[SN_MAX_AGGS][3], aisf[SN_MAX_AGGS][3];
  for (int a = 0; a < p->naggs; a++) {
    for (int f = 0; f < 3; f++) { aa[a][f] = 1.0; am[a][f] = 0.0; ab[a][f] = NULL; }
    if (p->aggs[a].kind == SN_AGG_COUNT_STAR) continue;
    for (int f = 0; f < p->aggs[a].nfactors; f++) {
      int c = p->aggs[a].factors[f].col;
      aa[a][f] = p->aggs[a].factors[f].add;
      am[a][f] = p->aggs[a].factors[f].mul;
      ab[a][f] = cols[c].base; aw[a][f] = cols[c].w; aisf[a][f] = cols[c].is_f;
      if (t->dtypes[c] == SN_TYPE_INT64) return 0;
    }
  }

  /* local accumulators */
  double *sums = (double *)calloc((size_t)nslots * p->naggs, 8);
  double *rcnt = (double *)calloc((size_t)nslots, 8);
  int32_t n = b->num_rows;
  (*rows_scanned) += n;

  for (int32_t r = 0; r < n; r++) {
    int ok = 1;
    for (int i = 0; i < p->npreds; i++) {
      double x;
      const uint8_t *bp = pbase[i];
      switch (pw[i]) {
        case 8: x = pisf[i] ? rd_f64(bp + (int64_t)r * 8) : 0.0; break;
        case 4: x = pisf[i] ? (double)rd_f32(bp + (int64_t)r * 4)
                            : (double)rd_i32(bp + (int64_t)r * 4); break;
        case 2: x = (double)rd_i16(bp + (int64_t)r * 2); break;
        default: x = (double)(int8_t)bp[r];
      }
      ok &= (x >= plo[i]) & (x <= phi[i]);
    }
    if (!ok) continue;
    int slot = 0;
    if (p->ngroup >= 1) slot = (uint16_t)rd_i16(gidx[0] + (int64_t)r * 2);
    if (p->ngroup == 2) slot = slot * gdictn[1] +
                               (uint16_t)rd_i16(gidx[1] + (int64_t)r * 2);
    rcnt[slot] += 1.0;
    double *srow = sums + (size_t)slot * p->naggs;
    for (int a = 0; a < p->naggs; a++) {
      double v = 1.0;
      for (int f = 0; f < 3; f++) {
        if (!ab[a][f]) { v *= aa[a][f] + 0.0; continue; }
        double x;
        const uint8_t *bp = ab[a][f];
        switch (aw[a][f]) {
          case 8: x = aisf[a][f] ? rd_f64(bp + (int64_t)r * 8) : 0.0; break;
          case 4: x = aisf[a][f] ? (double)rd_f32(bp + (int64_t)r * 4)
                                 : (double)rd_i32(bp + (int64_t)r * 4); break;
          case 2: x = (double)rd_i16(bp + (int64_t)r * 2); break;
          default: x = (double)(int8_t)bp[r];
        }
        v *= aa[a][f] + am[a][f] * x;
      }
      srow[a] += v;
    }
  }
  (*rows_passed) += 0;   /* recomputed below from rcnt */

  /* fold local slots into the group table */
  for (int slot = 0; slot < nslots; slot++) {
    if (rcnt[slot] == 0.0) continue;
    (*rows_passed) += (int64_t)rcnt[slot];
    char keys[SN_MAX_GROUPS][SN_KEY_MAX];
    uint8_t knull[SN_MAX_GROUPS];
    memset(knull, 0, sizeof(knull));
    int idx0 = p->ngroup >= 1 ? (p->ngroup == 2 ? slot / gdictn[1] : slot) : 0;
    int idx1 = p->ngroup == 2 ? slot % gdictn[1] : 0;
    int idxs[2] = { idx0, idx1 };
    for (int i = 0; i < p->ngroup; i++) {
      /* walk dict entries to entry idxs[i] ([int32 len][utf8]...) */
      const uint8_t *e = gdict_entries[i];
      for (int j = 0; j < idxs[i]; j++) e += 4 + rd_i32(e);
      int32_t L = rd_i32(e);
      if (L > SN_KEY_MAX - 1) L = SN_KEY_MAX - 1;
      memcpy(keys[i], e + 4, (size_t)L);
      keys[i][L] = 0;
    }
    sno_group *grp = gtab_get(g, (const char (*)[SN_KEY_MAX])keys, knull);
    if (!grp) { free(sums); free(rcnt); return 0; }
    grp->rowcount += rcnt[slot];
    for (int a = 0; a < p->naggs; a++) {
      if (p->aggs[a].kind == SN_AGG_COUNT_STAR) {
        grp->sums[a] += rcnt[slot];
        grp->counts[a] += rcnt[slot];
      } else {
        grp->sums[a] += sums[(size_t)slot * p->naggs + a];
        grp->counts[a] += rcnt[slot];
      }
    }
  }
  free(sums); free(rcnt);
  return 1;
}

static int group_cmp(const void *a, const void *b) {
  const sno_group *x = (const sno_group *)a, *y = (const sno_group *)b;
  for (int i = 0; i < SN_MAX_GROUPS; i++) {
    if (x->key_null[i] != y->key_null[i]) return x->key_null[i] - y->key_null[i];
    int c = strncmp(x->keys[i], y->keys[i], SN_KEY_MAX);
    if (c) return c;
  }
  return 0;
}

/* shared evaluation: fills the group table (sorted) + metrics
 * m = {rows_scanned, rows_passed, batches_seen, batches_skipped} */
static int32_t sno_run(sno_table *t, const sn_plan *p, int32_t nthreads,
                       sno_gtab *gout, int64_t m[4]) {
  if (!t || !p) return SN_ERR_BADARG;
  if (p->ngroup > SN_MAX_GROUPS || p->naggs > SN_MAX_AGGS || p->npreds > SN_MAX_PREDS)
    return SN_ERR_BADARG;
  int rc = SN_OK;
  const int join_group = p->join_dim != SN_JOIN_NONE &&
                         p->join_mode == SN_JOIN_GROUP && t->dim_hk != NULL;
  const int eff_ngroup = join_group ? 1 + p->ngroup : p->ngroup;
  if (eff_ngroup > SN_MAX_GROUPS) return SN_ERR_UNSUPPORTED;

  sno_gtab g;
  gtab_init(&g, eff_ngroup, p->naggs);
  int64_t rows_scanned = 0, rows_passed = 0, seen = 0, skipped = 0;

  if (nthreads <= 1) {
    for (int32_t bi = 0; bi < t->nbatches; bi++) {
      sno_batch *b = &t->batches[bi];
      seen++;
      /* stats skip — only for batches without deltas/deletes */
      if (b->stats && !b->has_deltas && !b->del) {
        sno_stats st;
        if (stats_parse(b->stats, b->stats_len, t->ncols, t->dtypes, &st) == SN_OK) {
          int skip = stats_skip(&st, p, t->dtypes, b->num_rows);
          stats_free(&st);
          if (skip) { skipped++; continue; }
        }
      }
      if (eval_batch_fast(t, b, p, &g, &rows_scanned, &rows_passed)) continue;
      if ((rc = eval_batch(t, b, p, &g, &rows_scanned, &rows_passed)) != SN_OK) break;
    }
  } else {
#ifdef _OPENMP
    /* timed CPU-baseline path: per-thread group tables merged afterwards */
    int nb = t->nbatches;
    sno_gtab *tg = (sno_gtab *)calloc((size_t)nthreads, sizeof(sno_gtab));
    int *trc = (int *)calloc((size_t)nthreads, sizeof(int));
    int64_t *tscan = (int64_t *)calloc((size_t)nthreads, 8);
    int64_t *tpass = (int64_t *)calloc((size_t)nthreads, 8);
    int64_t *tseen = (int64_t *)calloc((size_t)nthreads, 8);
    int64_t *tskip = (int64_t *)calloc((size_t)nthreads, 8);
    for (int i = 0; i < nthreads; i++) gtab_init(&tg[i], eff_ngroup, p->naggs);
#pragma omp parallel for schedule(dynamic) num_threads(nthreads)
    for (int bi = 0; bi < nb; bi++) {
      int tid = omp_get_thread_num();
      if (trc[tid] != SN_OK) continue;
      sno_batch *b = &t->batches[bi];
      tseen[tid]++;
      if (b->stats && !b->has_deltas && !b->del) {
        sno_stats st;
        if (stats_parse(b->stats, b->stats_len, t->ncols, t->dtypes, &st) == SN_OK) {
          int skip = stats_skip(&st, p, t->dtypes, b->num_rows);
          stats_free(&st);
          if (skip) { tskip[tid]++; continue; }
        }
      }
      if (eval_batch_fast(t, b, p, &tg[tid], &tscan[tid], &tpass[tid])) continue;
      int r = eval_batch(t, b, p, &tg[tid], &tscan[tid], &tpass[tid]);
      if (r != SN_OK) trc[tid] = r;
    }
    for (int i = 0; i < nthreads; i++) {
      if (trc[i] != SN_OK) rc = trc[i];
      rows_scanned += tscan[i]; rows_passed += tpass[i];
      seen += tseen[i]; skipped += tskip[i];
      for (int32_t gi = 0; gi < tg[i].n; gi++) {
        sno_group *src = &tg[i].groups[gi];
        sno_group *dst = gtab_get(&g, (const char (*)[SN_KEY_MAX])src->keys, src->key_null);
        if (!dst) { rc = SN_ERR_OVERFLOW; break; }
        dst->rowcount += src->rowcount;
        for (int a = 0; a < p->naggs; a++) {
          int k = p->aggs[a].kind;
          if (k == SN_AGG_MIN || k == SN_AGG_MAX) {
            if (src->counts[a] > 0)
              dst->sums[a] = dst->counts[a] > 0
                  ? (k == SN_AGG_MIN ? fmin(dst->sums[a], src->sums[a])
                                     : fmax(dst->sums[a], src->sums[a]))
                  : src->sums[a];
          } else {
            dst->sums[a] += src->sums[a];
          }
          dst->counts[a] += src->counts[a];
        }
      }
      gtab_free(&tg[i]);
    }
    free(tg); free(trc); free(tscan); free(tpass); free(tseen); free(tskip);
#else
    return SN_ERR_UNSUPPORTED;
#endif
  }
  if (rc != SN_OK) { gtab_free(&g); return rc; }

  /* keyless aggregate with zero rows still yields one row (Spark semantics) */
  if (eff_ngroup == 0 && g.n == 0) {
    char keys[SN_MAX_GROUPS][SN_KEY_MAX]; uint8_t knull[SN_MAX_GROUPS];
    memset(keys, 0, sizeof(keys)); memset(knull, 0, sizeof(knull));
    gtab_get(&g, (const char (*)[SN_KEY_MAX])keys, knull);
  }

  qsort(g.groups, (size_t)g.n, sizeof(sno_group), group_cmp);
  *gout = g;                     /* caller owns (gtab_free) */
  m[0] = rows_scanned; m[1] = rows_passed; m[2] = seen; m[3] = skipped;
  return SN_OK;
}

/* fill the agg value + null flag per Spark semantics */
static void sno_fill_val(const sn_agg *ag, const sno_group *grp, int a,
                         double *val, uint8_t *isnull) {
  if (ag->kind == SN_AGG_COUNT_STAR) { *val = grp->sums[a]; *isnull = 0; }
  else if (ag->kind == SN_AGG_AVG) {
    if (grp->counts[a] > 0) { *val = grp->sums[a] / grp->counts[a]; *isnull = 0; }
    else { *val = 0; *isnull = 1; }
  } else {
    if (grp->counts[a] > 0) { *val = grp->sums[a]; *isnull = 0; }
    else { *val = 0; *isnull = 1; }
  }
}

SNO_EXPORT int32_t sno_query(sno_table *t, const sn_plan *p, sn_result *out,
                             int32_t nthreads) {
  if (!out) return SN_ERR_BADARG;
  memset(out, 0, sizeof(*out));
  sno_gtab g;
  int64_t m[4] = { 0, 0, 0, 0 };
  int32_t rc = sno_run(t, p, nthreads, &g, m);
  if (rc != SN_OK) return rc;
  const int join_group = p->join_dim != SN_JOIN_NONE &&
                         p->join_mode == SN_JOIN_GROUP && t->dim_hk != NULL;
  const int eff_ngroup = join_group ? 1 + p->ngroup : p->ngroup;
  if (g.n > SN_MAX_GROUP_SLOTS) {     /* fixed result page: use sno_query_groups */
    gtab_free(&g);
    return SN_ERR_OVERFLOW;
  }
  out->nrows = g.n; out->ngroup = eff_ngroup; out->naggs = p->naggs;
  out->rows_scanned = m[0]; out->rows_passed = m[1];
  out->batches_seen = m[2]; out->batches_skipped = m[3];
  for (int32_t i = 0; i < g.n; i++) {
    sno_group *grp = &g.groups[i];
    for (int k = 0; k < eff_ngroup; k++) {
      strncpy(out->keys[i][k], grp->keys[k], SN_KEY_MAX);
      out->key_is_null[i][k] = grp->key_null[k];
    }
    for (int a = 0; a < p->naggs; a++)
      sno_fill_val(&p->aggs[a], grp, a, &out->vals[i][a], &out->val_is_null[i][a]);
  }
  gtab_free(&g);
  return SN_OK;
}

/* flat-array result export for group cardinalities beyond the sn_result
 * page (the oracle leg of the engine's sparse hash-aggregate / big-group
 * parity tests).  keys = [cap][SN_MAX_GROUPS][SN_KEY_MAX] char,
 * knull = [cap][SN_MAX_GROUPS], vals/vnull = [cap][SN_MAX_AGGS].
 * Returns the TOTAL group count (sorted; only min(total, cap) rows are
 * written) or a negative error. */
SNO_EXPORT int64_t sno_query_groups(sno_table *t, const sn_plan *p,
                                    int32_t nthreads, int64_t cap,
                                    char *keys, uint8_t *knull,
                                    double *vals, uint8_t *vnull) {
  if (!keys || !knull || !vals || !vnull || cap < 0) return SN_ERR_BADARG;
  sno_gtab g;
  int64_t m[4] = { 0, 0, 0, 0 };
  int32_t rc = sno_run(t, p, nthreads, &g, m);
  if (rc != SN_OK) return rc;
  int64_t n = g.n < cap ? g.n : cap;
  for (int64_t i = 0; i < n; i++) {
    sno_group *grp = &g.groups[i];
    for (int k = 0; k < SN_MAX_GROUPS; k++) {
      memcpy(keys + (i * SN_MAX_GROUPS + k) * SN_KEY_MAX, grp->keys[k], SN_KEY_MAX);
      knull[i * SN_MAX_GROUPS + k] = grp->key_null[k];
    }
    for (int a = 0; a < p->naggs; a++)
      sno_fill_val(&p->aggs[a], grp, a, &vals[i * SN_MAX_AGGS + a],
                   &vnull[i * SN_MAX_AGGS + a]);
  }
  int64_t total = g.n;
  gtab_free(&g);
  return total;
}

/* =======================================================================
 * Encoders (oracle-side writers for test vectors + golden fixtures)
 * ======================================================================= */
typedef struct { uint8_t *p; int64_t cap, off; int err; } wbuf;
static void wb_bytes(wbuf *w, const void *src, int64_t n) {
  if (w->err || w->off + n > w->cap) { w->err = 1; return; }
  memcpy(w->p + w->off, src, (size_t)n); w->off += n;
}
static void wb_i32(wbuf *w, int32_t v) { wb_bytes(w, &v, 4); }
static void wb_zero(wbuf *w, int64_t n) {
  if (w->err || w->off + n > w->cap) { w->err = 1; return; }
  memset(w->p + w->off, 0, (size_t)n); w->off += n;
}

/* write header: typeId + null bitset from validity (1=valid).  Returns count
 * of nulls. */
static int32_t wb_header(wbuf *w, int32_t type_id, const uint8_t *valid, int32_t count) {
  wb_i32(w, type_id);
  int32_t nnull = 0;
  if (valid) for (int32_t i = 0; i < count; i++) nnull += !valid[i];
  if (nnull == 0) { wb_i32(w, 0); return 0; }
  int32_t words = (count + 63) >> 6;
  wb_i32(w, words * 8);
  if (!w->err && w->off + (int64_t)words * 8 <= w->cap) {
    memset(w->p + w->off, 0, (size_t)words * 8);
    for (int32_t i = 0; i < count; i++)
      if (!valid[i]) bitset_set(w->p + w->off, i);
    w->off += (int64_t)words * 8;
  } else w->err = 1;
  return nnull;
}

SNO_EXPORT int64_t sno_encode(int32_t dtype, int32_t encoding, int32_t nullable,
    const void *values, const int32_t *str_lens, const uint8_t *valid,
    int32_t count, uint8_t *out, int64_t cap) {
  (void)nullable;
  wbuf w = { out, cap, 0, 0 };
  const uint8_t *v8 = (const uint8_t *)values;
  if (encoding == SN_ENC_UNCOMPRESSED) {
    wb_header(&w, 0, valid, count);
    if (dtype == SN_TYPE_STRING) {
      int64_t so = 0;
      for (int32_t i = 0; i < count; i++) {
        int32_t L = str_lens[i];
        if (!valid || valid[i]) { wb_i32(&w, L); wb_bytes(&w, v8 + so, L); }
        so += L;
      }
    } else {
      int wdt = type_width(dtype);
      for (int32_t i = 0; i < count; i++)
        if (!valid || valid[i]) wb_bytes(&w, v8 + (int64_t)i * wdt, wdt);
    }
  } else if (encoding == SN_ENC_RUNLENGTH) {
    if (dtype == SN_TYPE_INT8 || dtype == SN_TYPE_BOOL) return SN_ERR_UNSUPPORTED;
    wb_header(&w, 1, valid, count);
    int wdt = dtype == SN_TYPE_STRING ? 0 : type_width(dtype);
    /* runs over the non-null sequence */
    int32_t i = 0;
    int64_t so = 0; /* string byte offset */
    int64_t run_start_so = 0;
    int have = 0; int64_t curv = 0; int32_t curlen = 0, run = 0;
    for (i = 0; i <= count; i++) {
      int isval = (i < count) && (!valid || valid[i]);
      int64_t nv = 0; int32_t nl = 0; int64_t nso = so;
      if (i < count && dtype == SN_TYPE_STRING) { nl = str_lens[i]; so += nl; }
      if (isval && dtype != SN_TYPE_STRING) {
        switch (wdt) { case 2: nv = rd_i16(v8 + (int64_t)i*2); break;
                       case 4: nv = rd_i32(v8 + (int64_t)i*4); break;
                       default: nv = rd_i64(v8 + (int64_t)i*8); }
      }
      int same = have && isval &&
        (dtype == SN_TYPE_STRING
           ? (nl == curlen && memcmp(v8 + run_start_so, v8 + nso, (size_t)nl) == 0)
           : nv == curv);
      if (same) { run++; continue; }
      if (have && run > 0) {  /* flush */
        if (dtype == SN_TYPE_STRING) {
          wb_i32(&w, curlen); wb_bytes(&w, v8 + run_start_so, curlen); wb_i32(&w, run);
        } else {
          switch (wdt) { case 2: { int16_t t16=(int16_t)curv; wb_bytes(&w,&t16,2);} break;
                         case 4: { int32_t t32=(int32_t)curv; wb_bytes(&w,&t32,4);} break;
                         default: wb_bytes(&w,&curv,8); }
          wb_i32(&w, run);
        }
        have = 0; run = 0;
      }
      if (isval) { have = 1; run = 1; curv = nv; curlen = nl; run_start_so = nso; }
    }
  } else if (encoding == SN_ENC_DICTIONARY || encoding == SN_ENC_BIG_DICTIONARY) {
    int big = encoding == SN_ENC_BIG_DICTIONARY;
    wb_header(&w, big ? 3 : 2, valid, count);
    if (dtype == SN_TYPE_STRING) {
      /* build dict in first-occurrence order */
      int32_t dn = 0;
      int32_t *doff = (int32_t *)malloc(sizeof(int32_t) * (size_t)(count ? count : 1));
      int32_t *dlen = (int32_t *)malloc(sizeof(int32_t) * (size_t)(count ? count : 1));
      int32_t *idx = (int32_t *)malloc(sizeof(int32_t) * (size_t)(count ? count : 1));
      int64_t so = 0;
      for (int32_t i = 0; i < count; i++) {
        int32_t L = str_lens[i];
        if (valid && !valid[i]) { idx[i] = -1; so += L; continue; }
        int32_t j = 0;
        for (; j < dn; j++)
          if (dlen[j] == L && memcmp(v8 + doff[j], v8 + so, (size_t)L) == 0) break;
        if (j == dn) { doff[dn] = (int32_t)so; dlen[dn] = L; dn++; }
        idx[i] = j; so += L;
      }
      wb_i32(&w, dn);
      for (int32_t j = 0; j < dn; j++) { wb_i32(&w, dlen[j]); wb_bytes(&w, v8 + doff[j], dlen[j]); }
      /* index array holds only NON-NULL entries (NullableEncoder.writeIsNull
       * sets the bit and writes nothing, ColumnEncoding.scala:1239-1253) */
      for (int32_t i = 0; i < count; i++) {
        if (idx[i] < 0) continue;
        if (big) wb_i32(&w, idx[i]);
        else { int16_t x = (int16_t)idx[i]; wb_bytes(&w, &x, 2); }
      }
      free(doff); free(dlen); free(idx);
    } else if (dtype == SN_TYPE_INT32 || dtype == SN_TYPE_INT64) {
      int wdt = type_width(dtype);
      int32_t dn = 0;
      int64_t *dv = (int64_t *)malloc(8 * (size_t)(count ? count : 1));
      int32_t *idx = (int32_t *)malloc(4 * (size_t)(count ? count : 1));
      for (int32_t i = 0; i < count; i++) {
        if (valid && !valid[i]) { idx[i] = -1; continue; }
        int64_t x = wdt == 4 ? rd_i32(v8 + (int64_t)i*4) : rd_i64(v8 + (int64_t)i*8);
        int32_t j = 0;
        for (; j < dn; j++) if (dv[j] == x) break;
        if (j == dn) dv[dn++] = x;
        idx[i] = j;
      }
      wb_i32(&w, dn);
      for (int32_t j = 0; j < dn; j++) {
        if (wdt == 4) { int32_t x = (int32_t)dv[j]; wb_bytes(&w, &x, 4); }
        else wb_bytes(&w, &dv[j], 8);
      }
      for (int32_t i = 0; i < count; i++) {
        if (idx[i] < 0) continue;   /* nulls write no index entry */
        if (big) wb_i32(&w, idx[i]);
        else { int16_t x = (int16_t)idx[i]; wb_bytes(&w, &x, 2); }
      }
      free(dv); free(idx);
    } else return SN_ERR_UNSUPPORTED;
  } else if (encoding == SN_ENC_BOOLEAN_BITSET) {
    if (dtype != SN_TYPE_BOOL) return SN_ERR_UNSUPPORTED;
    int32_t nnull = 0;
    wb_header(&w, 4, valid, count);
    if (valid) for (int32_t i = 0; i < count; i++) nnull += !valid[i];
    int32_t nbits = count - nnull;
    int32_t words = (nbits + 63) >> 6;
    int64_t start = w.off;
    wb_zero(&w, (int64_t)words * 8);
    int32_t nnp = 0;
    for (int32_t i = 0; i < count && !w.err; i++) {
      if (valid && !valid[i]) continue;
      if (v8[i]) bitset_set(w.p + start, nnp);
      nnp++;
    }
  } else return SN_ERR_UNSUPPORTED;
  return w.err ? SN_ERR_NOMEM : w.off;
}

SNO_EXPORT int64_t sno_encode_delete(const int32_t *positions, int32_t n,
    int32_t num_base_rows, uint8_t *out, int64_t cap) {
  wbuf w = { out, cap, 0, 0 };
  wb_i32(&w, 0);               /* header for future use */
  wb_i32(&w, num_base_rows);
  wb_i32(&w, n);
  wb_bytes(&w, positions, (int64_t)n * 4);
  return w.err ? SN_ERR_NOMEM : w.off;
}

SNO_EXPORT int64_t sno_encode_delta(int32_t dtype, int32_t encoding, int32_t nullable,
    const int32_t *positions, int32_t n, int32_t num_base_rows,
    const void *values, const int32_t *str_lens, const uint8_t *valid,
    uint8_t *out, int64_t cap) {
  /* encode the full blob (header + body) to a temp buffer, then splice the
   * positions section between its null header and its body */
  int64_t tmp_cap = cap + 64;
  uint8_t *tmp = (uint8_t *)malloc((size_t)tmp_cap);
  int64_t blen = sno_encode(dtype, encoding, nullable, values, str_lens, valid, n, tmp, tmp_cap);
  if (blen < 0) { free(tmp); return blen; }
  /* split: [typeId][nullBytes][nullwords] | [body] */
  int32_t null_bytes = rd_i32(tmp + 4);
  int64_t hdr = 8 + null_bytes;
  wbuf w = { out, cap, 0, 0 };
  wb_bytes(&w, tmp, hdr);
  wb_i32(&w, num_base_rows);
  wb_i32(&w, n);
  wb_bytes(&w, positions, (int64_t)n * 4);
  int64_t pad = ((w.off + 7) & ~7LL) - w.off;
  wb_zero(&w, pad);
  wb_bytes(&w, tmp + hdr, blen - hdr);
  free(tmp);
  return w.err ? SN_ERR_NOMEM : w.off;
}

SNO_EXPORT int64_t sno_encode_stats(int32_t ncols, const int32_t *dtypes,
    int32_t batch_count_signed, const double *lower_d, const double *upper_d,
    const int64_t *lower_i, const int64_t *upper_i, const int32_t *null_counts,
    const uint8_t *has_bounds, uint8_t *out, int64_t cap) {
  int32_t num_fields = ncols * 3 + 1;
  int32_t null_words = (num_fields + 63) >> 6;
  int64_t need = (int64_t)null_words * 8 + (int64_t)num_fields * 8;
  if (cap < need) return SN_ERR_NOMEM;
  memset(out, 0, (size_t)need);
  uint8_t *bits = out;
  uint8_t *slots = out + (int64_t)null_words * 8;
  wr_i32(slots, batch_count_signed);
  for (int c = 0; c < ncols; c++) {
    int f_lo = 1 + c * 3, f_hi = 2 + c * 3, f_nc = 3 + c * 3;
    wr_i32(slots + (int64_t)f_nc * 8, null_counts ? null_counts[c] : 0);
    if (!has_bounds || !has_bounds[c]) { bitset_set(bits, f_lo); bitset_set(bits, f_hi); continue; }
    switch (dtypes[c]) {
      case SN_TYPE_DOUBLE: case SN_TYPE_FLOAT:
        wr_f64(slots + (int64_t)f_lo * 8, lower_d[c]);
        wr_f64(slots + (int64_t)f_hi * 8, upper_d[c]);
        break;
      case SN_TYPE_INT64:
        wr_i64(slots + (int64_t)f_lo * 8, lower_i[c]);
        wr_i64(slots + (int64_t)f_hi * 8, upper_i[c]);
        break;
      default:
        wr_i32(slots + (int64_t)f_lo * 8, (int32_t)lower_i[c]);
        wr_i32(slots + (int64_t)f_hi * 8, (int32_t)upper_i[c]);
    }
  }
  return need;
}

/* =======================================================================
 * Decode helper for round-trip tests
 * ======================================================================= */
SNO_EXPORT int32_t sno_decode(int32_t dtype, const uint8_t *blob, int64_t len,
    int32_t count, void *out_values, int64_t out_cap, int32_t *out_str_lens,
    uint8_t *out_valid) {
  sno_dec d;
  int rc = dec_init(&d, dtype, blob, len, 0, NULL, NULL);
  if (rc != SN_OK) return rc;
  uint8_t *ov = (uint8_t *)out_values;
  int64_t so = 0;
  for (int32_t ord = 0; ord < count; ord++) {
    int isnull = dec_null_advance(&d, ord);
    int32_t nnp = ord - d.nulls_before;
    if (out_valid) out_valid[ord] = (uint8_t)!isnull;
    if (isnull) { if (dtype == SN_TYPE_STRING && out_str_lens) out_str_lens[ord] = 0; continue; }
    if (dtype == SN_TYPE_STRING) {
      const uint8_t *s; int32_t sl;
      rc = dec_read_string(&d, nnp, &s, &sl);
      if (rc != SN_OK) { dec_free(&d); return rc; }
      if (s == NULL) { if (out_valid) out_valid[ord] = 0; if (out_str_lens) out_str_lens[ord] = 0; continue; }
      if (so + sl > out_cap) { dec_free(&d); return SN_ERR_NOMEM; }
      memcpy(ov + so, s, (size_t)sl); so += sl;
      if (out_str_lens) out_str_lens[ord] = sl;
    } else if (dtype == SN_TYPE_DOUBLE) {
      ((double *)ov)[ord] = dec_read_f64(&d, nnp);
    } else if (dtype == SN_TYPE_FLOAT) {
      ((float *)ov)[ord] = (float)dec_read_f64(&d, nnp);
    } else {
      int64_t x = dec_read_i64(&d, nnp);
      switch (dtype) {
        case SN_TYPE_INT32: ((int32_t *)ov)[ord] = (int32_t)x; break;
        case SN_TYPE_INT64: ((int64_t *)ov)[ord] = x; break;
        case SN_TYPE_INT16: ((int16_t *)ov)[ord] = (int16_t)x; break;
        case SN_TYPE_INT8: case SN_TYPE_BOOL: ov[ord] = (uint8_t)x; break;
      }
    }
  }
  dec_free(&d);
  return SN_OK;
}

/* expose table stats for tests */
SNO_EXPORT int32_t sno_table_nbatches(sno_table *t) { return t ? t->nbatches : -1; }
