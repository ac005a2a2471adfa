"""Round-2 surface widening, parity-tested against the oracle:
- RLE int64 columns on the GENERAL path (delete mask forces read_general) —
  the raw-bit run values must not be rounded (advisor regression)
- int64 aggregate factors (SUM(bigint) bit-exact under the stats-proven
  2^53 bound; int64 inside double-typed expressions within 1e-6)
- int-typed dictionary columns (materialized to plain bodies at put)
- group keys longer than SN_KEY_MAX-1 rejected loudly (never silently
  merged by truncation)
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


def assert_close(got, exp, rel=REL):
    assert abs(got - exp) <= rel * max(1.0, abs(exp)), (got, exp)


@pytest.mark.gpu
def test_rle_int64_general_path_with_deletes(eng):
    """RLE int64 + delete mask: the delete forces the general read path,
    whose RLE case must bitcast the raw-bit run values (SN_K_RLE_I64), not
    round the bit pattern (a value of 100 would read as 0)."""
    n = 120_000
    rng = np.random.default_rng(17)
    runs = np.repeat(rng.integers(1, 200, 3000), rng.integers(10, 90, 3000))[:n]
    i64 = (runs.astype(np.int64) * 5) + (1 << 58)   # beyond 2^53 exactness
    w = rng.random(n)
    dels = np.unique(rng.integers(0, n, n // 20)).astype(np.int32)
    dmask = se.encode_delete_mask(dels, n)
    cols = [po.encode(po.T_INT64, po.ENC_RLE, i64),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    t = eng.table_define("trle64", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 1, 0, n, cols, delete_mask=dmask)
    cut = int(np.median(i64))
    plan_kw = dict(preds=[dict(col=0, lo=cut, lo_strict=True)],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_INT64, po.T_DOUBLE])
    ot.add_batch(n, cols, delete_mask=dmask)
    orows = po.result_rows(ot.query(po.make_plan(**plan_kw)))
    assert grows[0][1][1] == orows[0][1][1]            # COUNT bit-exact
    assert_close(grows[0][1][0], orows[0][1][0])
    # numpy cross-check
    alive = np.ones(n, dtype=bool)
    alive[dels] = False
    m = alive & (i64 > cut)
    assert grows[0][1][1] == float(m.sum()) and m.sum() > 0


@pytest.mark.gpu
def test_int64_sum_bit_exact(eng):
    """SUM over an int64 column: bit-exact (stats prove every partial sum
    stays below 2^53); also int64 as a factor of a double expression."""
    n = 300_000
    rng = np.random.default_rng(23)
    v = rng.integers(-10**9, 10**9, n).astype(np.int64)
    d = rng.random(n)
    t = eng.table_define("ti64agg", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": v}, {"data": d}], n, batch_rows=70_000)
    plan_kw = dict(preds=[dict(col=0, lo=0)],
                   aggs=[("sum", [(0, 0.0, 1.0)]),                 # SUM(bigint)
                         ("sum", [(0, 0.0, 1.0), (1, 0.0, 1.0)]),  # i64 * double
                         ("avg", [(0, 0.0, 1.0)]),
                         ("count", [])])
    q = eng.query(abi.make_plan(table=t, **plan_kw))
    grows = q.rows()
    assert q.used_jit()        # i64 factors stay on the query-compiled path
    ot = po.OracleTable([po.T_INT64, po.T_DOUBLE])
    ot.add_batch(n, [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, v),
                     po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, d)])
    orows = po.result_rows(ot.query(po.make_plan(**plan_kw)))
    m = v >= 0
    assert grows[0][1][0] == orows[0][1][0] == float(v[m].sum())
    assert_close(grows[0][1][1], orows[0][1][1])
    assert_close(grows[0][1][2], orows[0][1][2])
    assert grows[0][1][3] == orows[0][1][3] == float(m.sum())


@pytest.mark.gpu
def test_int64_sum_rejects_unprovable_range(eng):
    """SUM(bigint) whose stats bound reaches 2^53 must fail loudly, not
    round silently (the reference's LongType sum is exact)."""
    n = 10_000
    v = np.full(n, 1 << 55, dtype=np.int64)
    t = eng.table_define("ti64big", [(abi.T_INT64, False)])
    eng.ingest_columns(t, [{"data": v}], n, batch_rows=n)
    with pytest.raises(se.EngineError) as ei:
        eng.query(abi.make_plan(table=t, aggs=[("sum", [(0, 0.0, 1.0)])]))
    assert "f64-exact" in str(ei.value)


@pytest.mark.gpu
@pytest.mark.parametrize("dt,pdt,vals", [
    ("i32", po.T_INT32, lambda rng, n: rng.integers(0, 40, n).astype(np.int32) * 7),
    ("i64", po.T_INT64, lambda rng, n: (rng.integers(0, 40, n).astype(np.int64) << 40) + 3),
])
def test_int_dictionary_columns(eng, dt, pdt, vals):
    """Int-typed dictionary columns (DictionaryEncoding over int32/int64):
    materialized to plain bodies at put; predicates exact (raw bits for i64)."""
    n = 150_000
    rng = np.random.default_rng(29)
    v = vals(rng, n)
    w = rng.random(n)
    abit = abi.T_INT32 if dt == "i32" else abi.T_INT64
    cols = [po.encode(pdt, po.ENC_DICT, v),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    t = eng.table_define("tdict_" + dt, [(abit, False), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 1, 0, n, cols)
    cut = int(np.median(v))
    plan_kw = dict(preds=[dict(col=0, hi=cut)],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([pdt, po.T_DOUBLE])
    ot.add_batch(n, cols)
    orows = po.result_rows(ot.query(po.make_plan(**plan_kw)))
    assert grows[0][1][1] == orows[0][1][1]
    assert_close(grows[0][1][0], orows[0][1][0])
    m = v <= cut
    assert grows[0][1][1] == float(m.sum()) and 0 < m.sum() < n


@pytest.mark.gpu
def test_int_dictionary_nullable_with_sentinel(eng):
    """Nullable int dictionary: nulls from the bitset AND from the
    index==numElements sentinel both fold into the materialized bitset."""
    n = 50_000
    rng = np.random.default_rng(31)
    v = rng.integers(0, 25, n).astype(np.int32)
    valid = (rng.random(n) > 0.1).astype(np.uint8)
    w = rng.random(n)
    cols = [po.encode(po.T_INT32, po.ENC_DICT, v, valid=valid),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    t = eng.table_define("tdictnull", [(abi.T_INT32, True), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 1, 0, n, cols)
    plan_kw = dict(preds=[dict(col=0, lo=10)],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_INT32, po.T_DOUBLE], nullable=[True, False])
    ot.add_batch(n, cols)
    orows = po.result_rows(ot.query(po.make_plan(**plan_kw)))
    assert grows[0][1][1] == orows[0][1][1]
    assert_close(grows[0][1][0], orows[0][1][0])
    m = (valid == 1) & (v >= 10)
    assert grows[0][1][1] == float(m.sum())


@pytest.mark.gpu
def test_long_group_key_rejected(eng):
    """A >47-byte dictionary entry on a group column fails the query loudly
    (SN_ERR_UNSUPPORTED) instead of silently merging truncated keys."""
    n = 1000
    longkey = b"X" * 60
    keys = [longkey if i % 2 else b"short" for i in range(n)]
    lens = np.array([len(k) for k in keys], dtype=np.int32)
    payload = b"".join(keys)
    w = np.ones(n)
    t = eng.table_define("tlongkey", [(abi.T_STRING, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": payload, "lens": lens}, {"data": w}], n,
                       batch_rows=n)
    with pytest.raises(se.EngineError) as ei:
        eng.query(abi.make_plan(table=t, group_cols=[0],
                                aggs=[("sum", [(1, 0.0, 1.0)])]))
    assert "key" in str(ei.value)
    # non-grouping queries over the same table still run
    rows = eng.query(abi.make_plan(table=t,
                                   aggs=[("count", [])])).rows()
    assert rows[0][1][0] == float(n)


@pytest.mark.gpu
def test_nullable_grouped_agg_inputs(eng):
    """Grouped aggregates over nullable inputs (previously rejected): rows
    with a null factor skip that aggregate but still count in COUNT(*) and
    rowcount (Spark Sum/Average semantics); AVG divides by the per-agg
    non-null count.  Routed through the per-agg-count (pac) layout."""
    n = 400_000
    rng = np.random.default_rng(71)
    keys = [b"G%02d" % v for v in rng.integers(0, 30, n)]
    a = rng.random(n)
    avalid = (rng.random(n) > 0.15).astype(np.uint8)
    b = rng.random(n)
    bvalid = (rng.random(n) > 0.4).astype(np.uint8)
    cols = [po.encode(po.T_STRING, po.ENC_DICT, keys),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, a, valid=avalid),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, b, valid=bvalid)]
    t = eng.table_define("tnullagg", [(abi.T_STRING, False), (abi.T_DOUBLE, True),
                                      (abi.T_DOUBLE, True)])
    for st in range(0, n, 100_000):
        en = min(n, st + 100_000)
        sub = [po.encode(po.T_STRING, po.ENC_DICT, keys[st:en]),
               po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, a[st:en],
                         valid=avalid[st:en]),
               po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, b[st:en],
                         valid=bvalid[st:en])]
        eng.batch_put(t, st, st // 100_000, en - st, sub)
    plan_kw = dict(group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]),
                         ("avg", [(1, 0.0, 1.0)]),
                         ("sum", [(1, 0.0, 1.0), (2, 0.0, 1.0)]),
                         ("avg", [(2, 0.0, 1.0)]),
                         ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_STRING, po.T_DOUBLE, po.T_DOUBLE],
                        nullable=[False, True, True])
    ot.add_batch(n, cols)
    orows = po.result_rows(ot.query(po.make_plan(**plan_kw)))
    assert_close_rows(grows, orows, count_aggs={4})
    # numpy cross-check of one group: sum skips nulls, count(*) does not
    g0 = np.array([k == b"G00" for k in keys])
    m = g0 & (avalid == 1)
    got = {k[0]: v for k, v in grows}
    assert abs(got["G00"][0] - a[m].sum()) <= REL * max(1.0, abs(a[m].sum()))
    assert got["G00"][4] == float(g0.sum())


@pytest.mark.gpu
def test_nullable_grouped_sparse_keys(eng):
    """pac layout on the sparse hash-aggregate path: int64 keys + nullable
    aggregate input."""
    n = 300_000
    rng = np.random.default_rng(73)
    keys = rng.integers(0, 4_000, n).astype(np.int64) * (1 << 35)
    w = rng.random(n)
    valid = (rng.random(n) > 0.2).astype(np.uint8)
    cols = [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, keys),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w, valid=valid)]
    t = eng.table_define("tnullsparse", [(abi.T_INT64, False), (abi.T_DOUBLE, True)])
    eng.batch_put(t, 1, 0, n, cols)
    plan_kw = dict(group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("avg", [(1, 0.0, 1.0)]),
                         ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_INT64, po.T_DOUBLE], nullable=[False, True])
    ot.add_batch(n, cols)
    orows = ot.query_groups(po.make_plan(**plan_kw), nthreads=16)
    assert_close_rows(grows, orows, count_aggs={2})


def assert_close_rows(grows, orows, count_aggs=()):
    assert len(grows) == len(orows)
    for (gk, gv), (ok_, ov) in zip(grows, orows):
        assert gk == ok_
        for i, (g, o) in enumerate(zip(gv, ov)):
            if i in count_aggs:
                assert g == o, (gk, i, g, o)
            elif o is None:
                assert g is None, (gk, i, g)
            else:
                assert abs(g - o) <= REL * max(1.0, abs(o)), (gk, i, g, o)


def test_oracle_int64_delta_exact_cpu():
    """Oracle-side: int64 update deltas keep exactness beyond 2^53."""
    n = 10_000
    base = np.arange(n, dtype=np.int64) + (1 << 57)
    pos = np.arange(0, n, 7, dtype=np.int32)
    newv = base[pos] + 3
    d1 = po.encode_delta(po.T_INT64, po.ENC_UNCOMPRESSED, pos, n, newv)
    t = po.OracleTable([po.T_INT64])
    t.add_batch(-n, [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, base)],
                deltas=[(d1, None)])
    cut = int(base[pos[5]] + 3)
    rows = po.result_rows(t.query(po.make_plan(
        preds=[dict(col=0, lo=cut, hi=cut)], aggs=[("count", [])])))
    merged = base.copy()
    merged[pos] = newv
    assert rows[0][1][0] == float((merged == cut).sum())


@pytest.mark.gpu
def test_engine_int64_delta_exact(eng):
    """Engine-side: int64 update-delta values travel as raw bits (patch
    materialization writes them back exactly)."""
    n = 20_000
    base = np.arange(n, dtype=np.int64) + (1 << 57)
    w = np.ones(n)
    pos = np.arange(0, n, 11, dtype=np.int32)
    newv = base[pos] + n + 5          # collision-free with unpatched values
    d1 = se.encode_update_delta(abi.T_INT64, pos, n, newv)
    cols = [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, base),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    t = eng.table_define("tdelta64", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 1, 0, -n, cols,
                  deltas=[(d1, None), (None, None)])
    cut = int(base[pos[7]] + n + 5)
    grows = eng.query(abi.make_plan(
        table=t, preds=[dict(col=0, lo=cut, hi=cut)],
        aggs=[("count", [])])).rows()
    merged = base.copy()
    merged[pos] = newv
    assert (merged == cut).sum() == 1    # collision-free by construction
    assert grows[0][1][0] == 1.0


@pytest.mark.gpu
def test_nullable_grouped_wide_pac_global_route(eng):
    """pac accumulator too wide for LDS (800 slots x 12 aggs -> ~160KB+):
    the shared routing predicate must send BOTH the kernel and the
    engine's zeroing to the global-atomic path."""
    n = 500_000
    nslots = 800
    rng = np.random.default_rng(79)
    keys = rng.integers(0, nslots, n).astype(np.int32)
    vals = [rng.random(n) for _ in range(3)]
    valids = [(rng.random(n) > 0.2).astype(np.uint8) for _ in range(3)]
    cols = [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, keys)] + \
        [po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, v, valid=vd)
         for v, vd in zip(vals, valids)]
    t = eng.table_define("tpacglob", [(abi.T_INT32, False)] +
                         [(abi.T_DOUBLE, True)] * 3)
    eng.batch_put(t, 1, 0, n, cols)
    aggs = []
    for c in range(1, 4):
        aggs += [("sum", [(c, 0.0, 1.0)]), ("avg", [(c, 0.0, 1.0)]),
                 ("sum", [(c, 1.0, 2.0)]), ("avg", [(c, 2.0, -1.0)])]
    grows = eng.query(abi.make_plan(table=t, group_cols=[0],
                                    aggs=aggs)).rows()
    ot = po.OracleTable([po.T_INT32] + [po.T_DOUBLE] * 3,
                        nullable=[False, True, True, True])
    ot.add_batch(n, cols)
    orows = ot.query_groups(po.make_plan(group_cols=[0], aggs=aggs),
                            nthreads=8)
    assert len(grows) == len(orows)
    for (gk, gv), (ok_, ov) in zip(grows, orows):
        assert gk == ok_
        for g, o in zip(gv, ov):
            if o is None:
                assert g is None
            else:
                assert abs(g - o) <= 1e-6 * max(1.0, abs(o)), (gk, g, o)


@pytest.mark.gpu
def test_varwidth_string_columns_transcoded(eng):
    """Uncompressed (non-dictionary) string bodies — [len][bytes] sequential,
    no random access — transcode to Dictionary at put, so grouping and
    equality pushdown work on them (previously SN_ERR_UNSUPPORTED)."""
    n = 120_000
    rng = np.random.default_rng(137)
    words = [b"alpha", b"beta", b"gamma", b"delta", b"epsilon"]
    keys = [words[v] for v in rng.integers(0, 5, n)]
    w = rng.random(n)
    valid = (rng.random(n) > 0.1).astype(np.uint8)
    cols = [po.encode(po.T_STRING, po.ENC_UNCOMPRESSED, keys, valid=valid),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    t = eng.table_define("tvarstr", [(abi.T_STRING, True), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 1, 0, n, cols)
    # group by the var-width string column
    plan_kw = dict(group_cols=[0], aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_STRING, po.T_DOUBLE], nullable=[True, False])
    ot.add_batch(n, cols)
    orows = po.result_rows(ot.query(po.make_plan(**plan_kw)))
    assert len(grows) == len(orows) == 6       # 5 words + NULL key group
    for (gk, gv), (ok_, ov) in zip(grows, orows):
        assert gk == ok_ and gv[1] == ov[1]
        assert abs(gv[0] - ov[0]) <= REL * max(1.0, abs(ov[0]))
    # equality pushdown on the transcoded column
    q = eng.query(abi.make_plan(table=t, preds=[dict(col=0, eq=b"gamma")],
                                aggs=[("count", [])]))
    m = np.array([k == b"gamma" for k in keys]) & (valid == 1)
    assert q.rows()[0][1][0] == float(m.sum())


@pytest.mark.gpu
def test_order_by_topk_epilogue(eng):
    """ORDER BY <agg> [DESC] LIMIT k (SnappySortExec/TakeOrderedAndProject):
    reorders finalized groups by aggregate value, NULLs last, stable ties,
    truncates to k; agg_idx=-1 restores key order."""
    n = 500_000
    rng = np.random.default_rng(139)
    keys = rng.integers(0, 3_000, n).astype(np.int64) * 7919   # sparse path
    w = rng.random(n)
    t = eng.table_define("ttopk", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": w}], n, batch_rows=100_000)
    q = eng.query(abi.make_plan(table=t, group_cols=[0],
                                aggs=[("sum", [(1, 0.0, 1.0)]),
                                      ("count", [])]))
    q.wait()
    # numpy reference: top-10 groups by sum desc
    import collections
    sums = collections.defaultdict(float)
    cnts = collections.defaultdict(int)
    for kk, vv in zip(keys, w):
        sums[int(kk)] += vv
        cnts[int(kk)] += 1
    top = sorted(sums.items(), key=lambda kv: -kv[1])[:10]
    q.order_by(0, descending=True, k=10)
    rows = q.rows()
    assert len(rows) == 10
    for (gk, gv), (ek, ev) in zip(rows, top):
        assert int(gk[0]) == ek
        assert abs(gv[0] - ev) <= 1e-6 * max(1.0, abs(ev))
        assert gv[1] == float(cnts[ek])
    # ascending with count tie-breaking stability + key-order restore
    q2 = eng.query(abi.make_plan(table=t, group_cols=[0],
                                 aggs=[("count", [])]))
    q2.wait()
    q2.order_by(0, descending=False, k=5)
    asc = q2.rows()
    counts_sorted = sorted(cnts.values())
    assert [v[0] for _, v in asc] == [float(c) for c in counts_sorted[:5]]


@pytest.mark.gpu
def test_mutate_after_raw_ingest(eng):
    """UPDATE/DELETE against a batch that arrived through the f2 RAW path:
    patch materialization writes into the device-built body and the
    device-computed stats are superseded."""
    n = 30_000
    rng = np.random.default_rng(149)
    v = rng.random(n)
    t = eng.table_define("trawmut", [(abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": v}], n, batch_rows=n)   # uuid 0, bucket 0
    pos = np.arange(0, n, 13, dtype=np.int32)
    newv = v[pos] + 10.0
    d1 = se.encode_update_delta(abi.T_DOUBLE, pos, n, newv)
    dels = np.arange(0, n, 97, dtype=np.int32)
    eng.batch_mutate(t, 0, 0, deltas=[(d1, None)],
                     delete_mask=se.encode_delete_mask(dels, n))
    merged = v.copy()
    merged[pos] = newv
    alive = np.ones(n, dtype=bool)
    alive[dels] = False
    q = eng.query(abi.make_plan(table=t,
                                aggs=[("sum", [(0, 0.0, 1.0)]),
                                      ("max", [(0, 0.0, 1.0)]),
                                      ("count", [])]))
    rows = q.rows()
    exp = merged[alive].sum()
    assert abs(rows[0][1][0] - exp) <= 1e-9 * max(1.0, exp)
    assert rows[0][1][1] == merged[alive].max()
    assert rows[0][1][2] == float(alive.sum())


@pytest.mark.gpu
def test_duplicate_group_column_rejected(eng):
    """GROUP BY a, a: one device column slot cannot carry two premultiplied
    dictionary images — rejected loudly (fuzzer-found silent wrong keys)."""
    n = 10_000
    rng = np.random.default_rng(401)
    m = [b"A" if x else b"B" for x in rng.integers(0, 2, n)]
    w = np.ones(n)
    t = eng.table_define("tdupg", [(abi.T_STRING, False), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 0, 0, n, [po.encode(po.T_STRING, po.ENC_DICT, m),
                               po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)])
    with pytest.raises(se.EngineError):
        eng.query(abi.make_plan(table=t, group_cols=[0, 0],
                                aggs=[("count", [])]))


@pytest.mark.gpu
def test_short_chunk_lds_tail_not_poisoned(eng):
    """Regression for the fuzzer-found NaN: a keyless MIN/MAX+AVG plan over
    a table whose last tile is a SHORT chunk (rows % CHUNK != 0), run right
    after sparse MIN/MAX queries that park ord-ident NaN bit patterns in
    LDS.  Pre-fix, the slot-predicated fma accumulators computed
    fma(0, NaN-from-raw-LDS, sum) and silently poisoned the sums."""
    rng = np.random.default_rng(409)
    # step 1: sprinkle ord idents across every CU's LDS (sparse mm queries)
    ks = rng.integers(0, 50_000, 2_000_000).astype(np.int64) * (1 << 30)
    vs = rng.random(2_000_000)
    ts = eng.table_define("tpoison", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(ts, [{"data": ks}, {"data": vs}], len(ks),
                       batch_rows=500_000)
    for _ in range(3):
        eng.query(abi.make_plan(table=ts, group_cols=[0],
                                aggs=[("min", [(1, 0.0, 1.0)]),
                                      ("max", [(1, 0.0, 1.0)]),
                                      ("count", [])])).wait()
    # step 2: keyless pac plan over short-tailed batches (n % 1024 != 0)
    n = 33_007
    c0 = rng.integers(-5_000, 5_000, n).astype(np.int32)
    c1 = rng.integers(-300, 300, n).astype(np.int16)
    t = eng.table_define("tshort", [(abi.T_INT32, False), (abi.T_INT16, False)])
    eng.batch_put(t, 0, 0, n, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, c0),
                               po.encode(po.T_INT16, po.ENC_UNCOMPRESSED, c1)])
    aggs = [("avg", [(1, 0.0, 2.0), (0, 2.0, 1.0)]),
            ("max", [(1, -1.0, 1.0)]),
            ("avg", [(1, 2.0, 0.5)]),
            ("count", [])]
    for _ in range(3):
        rows = eng.query(abi.make_plan(table=t, aggs=aggs)).rows()
        vals = rows[0][1]
        assert all(v is not None and np.isfinite(v) for v in vals), vals
        assert vals[3] == float(n)
        assert abs(vals[2] - np.mean(2.0 + 0.5 * c1)) <= 1e-9 * 10
