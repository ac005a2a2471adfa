"""Sparse-key open-address hash aggregate (the ByteBufferHashMap /
SHAMapAccessor analogue, ByteBufferHashMap.scala:140-183,
SHAMapAccessor.scala:716-830): integer group keys WITHOUT dense-slot
structure — int64 keys, int32 spans beyond the dense 2^20 cap — probed
against an HBM open-address table (atomicCAS insert, device compaction).

Parity: GPU engine vs the oracle's own open hash table (sno_query_groups),
plus numpy cross-checks.  Also covers the variable-capacity partial blocks
(sn_query_partials_sharded2) that let >1024-group group-bys run the
key-sharded multi-GPU exchange — the north-star "hash table overflows one
GPU" demo.
"""
import numpy as np
import pytest

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se

REL = 1e-6


@pytest.fixture
def eng():
    e = se.Engine(device=0)
    yield e
    e.close()


def assert_rows_match(grows, orows, count_aggs=()):
    assert len(grows) == len(orows)
    for (gk, gv), (ok_, ov) in zip(grows, orows):
        assert gk == ok_
        for a, (g, o) in enumerate(zip(gv, ov)):
            if a in count_aggs:
                assert g == o, (gk, a, g, o)
            elif o is None:
                assert g is None
            else:
                assert abs(g - o) <= REL * max(1.0, abs(o)), (gk, a, g, o)


def test_oracle_open_hash_beyond_page_cpu():
    """Oracle-side: 20K distinct groups through the growing open hash
    table, flat export, against a numpy groupby."""
    n = 400_000
    rng = np.random.default_rng(41)
    keys = rng.integers(0, 20_000, n).astype(np.int64) * 1_000_003
    w = rng.random(n)
    t = po.OracleTable([po.T_INT64, po.T_DOUBLE])
    t.add_batch(n, [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, keys),
                    po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)])
    rows = t.query_groups(po.make_plan(group_cols=[0],
                                       aggs=[("sum", [(1, 0.0, 1.0)]),
                                             ("count", [])]))
    uk = np.unique(keys)
    assert len(rows) == len(uk)
    got = {k[0]: v for k, v in rows}
    for kv in uk[:50]:
        m = keys == kv
        assert got[str(kv)][1] == float(m.sum())
        assert abs(got[str(kv)][0] - w[m].sum()) <= REL * max(1.0, w[m].sum())


@pytest.mark.gpu
def test_sparse_int64_keys_100k_groups(eng):
    """The VERDICT done-criterion: grouped query over int64 keys with 10^5+
    distinct sparse values, GPU vs oracle."""
    n = 4_000_000
    ndistinct = 120_000
    rng = np.random.default_rng(43)
    universe = (rng.integers(-2**62, 2**62, ndistinct).astype(np.int64))
    universe[0] = -1                       # the hash sentinel as a REAL key
    universe[1] = 0
    keys = universe[rng.integers(0, ndistinct, n)]
    w = rng.random(n)
    t = eng.table_define("tsparse64", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": w}], n, batch_rows=500_000)
    plan_kw = dict(group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    q = eng.query(abi.make_plan(table=t, **plan_kw))
    grows = q.rows()
    assert len(grows) == len(np.unique(keys))
    ot = po.OracleTable([po.T_INT64, po.T_DOUBLE])
    for st in range(0, n, 500_000):
        en = min(n, st + 500_000)
        ot.add_batch(en - st,
                     [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, keys[st:en]),
                      po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w[st:en])])
    orows = ot.query_groups(po.make_plan(**plan_kw), nthreads=32)
    assert_rows_match(grows, orows, count_aggs={1})


@pytest.mark.gpu
def test_sparse_int32_wide_span(eng):
    """int32 keys whose stats span exceeds the dense 2^20 cap fall back to
    the hash aggregate (previously SN_ERR_UNSUPPORTED)."""
    n = 1_000_000
    rng = np.random.default_rng(47)
    keys = (rng.integers(0, 30_000, n).astype(np.int32) * 997 - 10**7)
    w = rng.random(n)
    t = eng.table_define("tsparse32", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": w}], n, batch_rows=250_000)
    plan_kw = dict(group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("avg", [(1, 0.0, 1.0)]),
                         ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_INT32, po.T_DOUBLE])
    ot.add_batch(n, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, keys),
                     po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)])
    orows = ot.query_groups(po.make_plan(**plan_kw), nthreads=32)
    assert_rows_match(grows, orows, count_aggs={2})


@pytest.mark.gpu
def test_sparse_two_int32_keys_packed(eng):
    """Two int32 key columns pack into one 64-bit hash key (reversible);
    results carry both key columns as decimal text like the dense path."""
    n = 800_000
    rng = np.random.default_rng(53)
    k0 = (rng.integers(0, 300, n).astype(np.int32) * 12347 - 5 * 10**6)
    k1 = rng.integers(-70, 70, n).astype(np.int32) * 10**6
    w = rng.random(n)
    t = eng.table_define("tsparse2k", [(abi.T_INT32, False), (abi.T_INT32, False),
                                       (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": k0}, {"data": k1}, {"data": w}], n,
                       batch_rows=200_000)
    plan_kw = dict(group_cols=[0, 1],
                   aggs=[("sum", [(2, 0.0, 1.0)]), ("count", [])],
                   preds=[dict(col=2, is_double=True, lo=0.25)])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_INT32, po.T_INT32, po.T_DOUBLE])
    ot.add_batch(n, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, k0),
                     po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, k1),
                     po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)])
    orows = ot.query_groups(po.make_plan(**plan_kw), nthreads=32)
    assert_rows_match(grows, orows, count_aggs={1})


@pytest.mark.gpu
def test_sparse_q1_style_aggregate_dedup(eng):
    """Sparse keys with the Q1 aggregate shape (shared expressions dedupe;
    COUNT(*) folds into rowcount) — exercises agg_map on the sparse path."""
    n = 600_000
    rng = np.random.default_rng(59)
    keys = rng.integers(0, 5_000, n).astype(np.int64) * (1 << 33)
    a = rng.random(n)
    b = rng.random(n)
    t = eng.table_define("tsparseq1", [(abi.T_INT64, False), (abi.T_DOUBLE, False),
                                       (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": a}, {"data": b}], n,
                       batch_rows=150_000)
    plan_kw = dict(group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]),
                         ("avg", [(1, 0.0, 1.0)]),
                         ("sum", [(1, 0.0, 1.0), (2, 1.0, -1.0)]),
                         ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_INT64, po.T_DOUBLE, po.T_DOUBLE])
    ot.add_batch(n, [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, keys),
                     po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, a),
                     po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, b)])
    orows = ot.query_groups(po.make_plan(**plan_kw), nthreads=32)
    assert_rows_match(grows, orows, count_aggs={3})


@pytest.mark.gpu
def test_key_sharded_a2a_20k_groups_split_merge(eng):
    """The >1024-group multi-GPU overflow exchange on one device: two shard
    engines (rank 0/1) each export variable-capacity key-sharded blocks
    (sn_query_partials_sharded2); cross-merging the blocks reproduces the
    unsharded result — the north-star overflow group-by demo's mechanics."""
    n = 1_000_000
    ngroups = 20_000
    rng = np.random.default_rng(61)
    keys = rng.integers(0, ngroups, n).astype(np.int64) * 1_000_003 - 4 * 10**9
    w = rng.random(n)

    def load(e2, shard=None):
        t2 = e2.table_define("t", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
        bi = 0
        for st in range(0, n, 100_000):
            en = min(n, st + 100_000)
            e2.ingest_columns(t2, [{"data": keys[st:en]}, {"data": w[st:en]}],
                              en - st, batch_rows=100_000, first_bucket=bi)
            bi += 1
        return t2
    plan_kw = dict(group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])

    t_all = load(eng)
    ref_rows = eng.query(abi.make_plan(table=t_all, **plan_kw)).rows()
    assert len(ref_rows) == len(np.unique(keys)) > 1024

    e0 = se.Engine(device=0, shard_rank=0, shard_count=2)
    e1 = se.Engine(device=0, shard_rank=1, shard_count=2)
    try:
        q0 = eng and e0.query(abi.make_plan(table=load(e0), **plan_kw))
        q1 = e1.query(abi.make_plan(table=load(e1), **plan_kw))
        cap = max(1024, q0.num_groups(), q1.num_groups())
        bb = q0.partial_bytes(cap)
        assert bb == q1.partial_bytes(cap)
        s0 = q0.partials_sharded(2, cap)    # [dest][bb]
        s1 = q1.partials_sharded(2, cap)
        # rank 0 merges column 0 of both; rank 1 merges column 1
        blocks0 = np.concatenate([s0[0], s1[0]])
        blocks1 = np.concatenate([s0[1], s1[1]])
        q0.merge_host(np.ascontiguousarray(blocks0), bb, 2)
        q1.merge_host(np.ascontiguousarray(blocks1), bb, 2)
        merged = sorted(q0.rows() + q1.rows())
        # sharded partials sum in a different order than the unsharded
        # reference: counts/keys exact, double sums within 1e-9 relative
        assert_rows_match(merged, sorted(ref_rows), count_aggs={1})
    finally:
        e0.close()
        e1.close()


@pytest.mark.gpu
def test_sparse_with_deletes_general_path(eng):
    """Sparse hash aggregate over a batch with a delete mask (general
    conversion path feeding the probe)."""
    n = 200_000
    rng = np.random.default_rng(67)
    keys = rng.integers(0, 3_000, n).astype(np.int64) * (1 << 30)
    w = rng.random(n)
    dels = np.unique(rng.integers(0, n, n // 10)).astype(np.int32)
    cols = [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, keys),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    dmask = se.encode_delete_mask(dels, n)
    t = eng.table_define("tsparsedel", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 1, 0, n, cols, delete_mask=dmask)
    plan_kw = dict(group_cols=[0], aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_INT64, po.T_DOUBLE])
    ot.add_batch(n, cols, delete_mask=dmask)
    orows = ot.query_groups(po.make_plan(**plan_kw), nthreads=16)
    assert_rows_match(grows, orows, count_aggs={1})


@pytest.mark.gpu
def test_sparse_nullable_int64_keys(eng):
    """Nullable int64 group keys (single-column): NULL keys form their own
    group (Spark GROUP BY null semantics), accumulated in the dedicated
    null row of the open-address table."""
    n = 250_000
    rng = np.random.default_rng(151)
    keys = rng.integers(0, 2_000, n).astype(np.int64) * (1 << 32)
    valid = (rng.random(n) > 0.15).astype(np.uint8)
    w = rng.random(n)
    cols = [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, keys, valid=valid),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    t = eng.table_define("tsparsenull", [(abi.T_INT64, True), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 1, 0, n, cols)
    plan_kw = dict(group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_INT64, po.T_DOUBLE], nullable=[True, False])
    ot.add_batch(n, cols)
    orows = ot.query_groups(po.make_plan(**plan_kw), nthreads=16)
    assert_rows_match(grows, orows, count_aggs={1})
    # the NULL group exists and counts exactly the invalid rows
    nulls = [v for k, v in grows if k[0] is None]
    assert len(nulls) == 1 and nulls[0][1] == float((valid == 0).sum())


@pytest.mark.gpu
def test_radix_glob_fallback_parity(eng, monkeypatch):
    """The radix pass-2 GLOBAL-segment variant (accumulator rows too wide
    for the LDS table — cap 2^24 with 4 aggregates in production, forced
    here via SN_RADIX_GLOB so it parity-checks at test size): min/max +
    sums through per-partition segments of the global table, then the
    regular k_hash_compact readback."""
    monkeypatch.setenv("SN_RADIX_GLOB", "1")
    n = 2_000_000
    ndistinct = 150_000
    rng = np.random.default_rng(211)
    universe = rng.integers(-2**62, 2**62, ndistinct).astype(np.int64)
    universe[0] = -1                       # sentinel as a REAL key
    keys = universe[rng.integers(0, ndistinct, n)]
    w = rng.random(n)
    v = rng.random(n) * 100 - 50
    t = eng.table_define("tradixg", [(abi.T_INT64, False), (abi.T_DOUBLE, False),
                                     (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": w}, {"data": v}], n,
                       batch_rows=500_000)
    plan_kw = dict(group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("min", [(2, 0.0, 1.0)]),
                         ("max", [(2, 0.0, 1.0)]), ("count", [])])
    q = eng.query(abi.make_plan(table=t, **plan_kw))
    grows = q.rows()
    assert q.used_jit()                    # radix pass 1 ran compiled
    ot = po.OracleTable([po.T_INT64, po.T_DOUBLE, po.T_DOUBLE])
    for st in range(0, n, 500_000):
        en = min(n, st + 500_000)
        ot.add_batch(en - st,
                     [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, keys[st:en]),
                      po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w[st:en]),
                      po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, v[st:en])])
    orows = ot.query_groups(po.make_plan(**plan_kw), nthreads=32)
    assert_rows_match(grows, orows, count_aggs={3})
