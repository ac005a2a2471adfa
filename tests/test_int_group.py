"""Integer group-by keys (stats-ranged direct slots — the
DictionaryOptimizedMapAccessor direct-slot idea applied to int columns;
reference SHAMap accepts any fixed-width key, SHAMapAccessor.scala:1106-1140).
Keys surface as decimal text in results and partial blocks."""
import numpy as np
import pytest

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se


def expected(keys, vals, mask=None):
    exp = {}
    m = np.ones(len(keys), bool) if mask is None else mask
    for k, v in zip(np.asarray(keys)[m].tolist(), np.asarray(vals)[m].tolist()):
        s, c = exp.get(str(k), (0.0, 0))
        exp[str(k)] = (s + v, c + 1)
    return exp


def check(rows, exp):
    assert len(rows) == len(exp)
    for gk, gv in rows:
        k = gk[0] if isinstance(gk, tuple) else gk
        s, c = exp[k]
        assert gv[1] == float(c)
        assert abs(gv[0] - s) <= 1e-9 * max(1.0, abs(s))


def test_oracle_int32_group_vs_numpy():
    n = 100_000
    rng = np.random.default_rng(71)
    keys = rng.integers(-5, 45, n).astype(np.int32)
    vals = rng.random(n)
    t = po.OracleTable([po.T_INT32, po.T_DOUBLE])
    t.add_batch(n, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, keys),
                    po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, vals)])
    rows = po.result_rows(t.query(po.make_plan(
        group_cols=[0], aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])))
    check(rows, expected(keys, vals))


@pytest.mark.gpu
def test_engine_int32_group(rng_seed=72):
    n = 400_000
    rng = np.random.default_rng(rng_seed)
    keys = rng.integers(100, 400, n).astype(np.int32)
    vals = rng.random(n)
    eng = se.Engine(device=0)
    t = eng.table_define("tig", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": vals}], n, batch_rows=60_000)
    q = eng.query(abi.make_plan(table=t, group_cols=[0],
                                aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])]))
    rows = q.rows()
    assert q.used_jit()
    check(rows, expected(keys, vals))
    eng.close()


@pytest.mark.gpu
def test_engine_mixed_dict_and_int16_keys():
    """two-column key: dictionary string x int16, with a filter."""
    n = 300_000
    rng = np.random.default_rng(73)
    s = [b"R%d" % v for v in rng.integers(0, 3, n)]
    k16 = rng.integers(-8, 8, n).astype(np.int16)
    vals = rng.random(n)
    eng = se.Engine(device=0)
    t = eng.table_define("tmk", [(abi.T_STRING, False), (abi.T_INT16, False),
                                 (abi.T_DOUBLE, False)])
    for bi, st in enumerate(range(0, n, 60_000)):
        en = min(n, st + 60_000)
        blobs = [po.encode(po.T_STRING, po.ENC_DICT, s[st:en]),
                 po.encode(po.T_INT16, po.ENC_UNCOMPRESSED, k16[st:en]),
                 po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, vals[st:en])]
        ks, vs = k16[st:en], vals[st:en]
        stats = po.encode_stats([po.T_STRING, po.T_INT16, po.T_DOUBLE], en - st,
                                [0, int(ks.min()), float(vs.min())],
                                [0, int(ks.max()), float(vs.max())])
        eng.batch_put(t, bi, bi, en - st, blobs, stats=stats)
    plan = abi.make_plan(table=t, preds=[dict(col=2, is_double=True, lo=0.25)],
                         group_cols=[0, 1],
                         aggs=[("sum", [(2, 0.0, 1.0)]), ("count", [])])
    q = eng.query(plan)
    rows = q.rows()
    m = vals >= 0.25
    exp = {}
    for sv, kv, vv in zip(np.array(s)[m], k16[m].tolist(), vals[m].tolist()):
        key = (sv.decode(), str(kv))
        acc = exp.get(key, (0.0, 0))
        exp[key] = (acc[0] + vv, acc[1] + 1)
    assert len(rows) == len(exp)
    for gk, gv in rows:
        sgot, cgot = exp[(gk[0], gk[1])]
        assert gv[1] == float(cgot)
        assert abs(gv[0] - sgot) <= 1e-9 * max(1.0, abs(sgot))
    eng.close()


@pytest.mark.gpu
def test_engine_int_group_high_cardinality_and_partials():
    """600-wide int key range (LDS-mode JIT) + sharded partial merge."""
    n = 500_000
    rng = np.random.default_rng(74)
    keys = rng.integers(1000, 1600, n).astype(np.int32)
    vals = rng.random(n)
    eng = se.Engine(device=0)
    t = eng.table_define("tih", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": vals}], n, batch_rows=50_000)
    plan = abi.make_plan(table=t, group_cols=[0],
                         aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    q = eng.query(plan)
    direct = q.rows()
    assert q.used_jit()
    check(direct, expected(keys, vals))
    block = q.partials_host()
    q.merge_host(block, len(block), 1)
    assert q.rows() == direct
    eng.close()


@pytest.mark.gpu
def test_big_group_cardinality_20k(rng_seed=75):
    """>1024 dense group slots (global-atomic accumulate path) with paged
    results: 20K distinct int keys vs a numpy model; partial export must
    refuse loudly (the overflow exchange is round-2)."""
    n = 2_000_000
    rng = np.random.default_rng(rng_seed)
    keys = rng.integers(0, 20_000, n).astype(np.int32)
    vals = rng.random(n)
    eng = se.Engine(device=0)
    t = eng.table_define("tbig", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": vals}], n,
                       batch_rows=250_000)
    q = eng.query(abi.make_plan(table=t, group_cols=[0],
                                aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])]))
    rows = q.rows()
    exp = expected(keys, vals)
    assert len(rows) == len(exp)
    check(rows, exp)
    try:
        q.partials_host()
        raise AssertionError("partial export must refuse > 1024 groups")
    except se.EngineError:
        pass
    eng.close()
