"""Broadcast-dimension join (config 4, HashJoinExec semantics):
CPU oracle vs a direct numpy model, and the GPU engine vs the oracle."""
import numpy as np
import pytest

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se

RNG = np.random.default_rng(99)


def make_fact(n=200_000, keyspace=10_000):
    keys = RNG.integers(0, keyspace, n).astype(np.int32)
    measure = RNG.random(n) * 100
    return keys, measure


def make_dim(keyspace=10_000, frac=0.3, nattrs=5):
    dk = np.sort(RNG.choice(keyspace, size=int(keyspace * frac),
                            replace=False)).astype(np.int64)
    attrs = [b"REGION_%d" % (RNG.integers(0, nattrs)) for _ in dk]
    return dk, attrs


def oracle_table(keys, measure):
    t = po.OracleTable([po.T_INT32, po.T_DOUBLE])
    for s in range(0, len(keys), 60_000):
        e = min(len(keys), s + 60_000)
        t.add_batch(e - s, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, keys[s:e]),
                            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, measure[s:e])])
    return t


def test_oracle_semi_join_vs_numpy():
    keys, measure = make_fact()
    dk, attrs = make_dim()
    t = oracle_table(keys, measure)
    t.set_dim(dk, attrs)
    plan = po.make_plan(aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
                        join=dict(dim=0, fact_col=0))
    rows = po.result_rows(t.query(plan))
    m = np.isin(keys, dk)
    assert rows[0][1][1] == float(m.sum())
    assert abs(rows[0][1][0] - measure[m].sum()) < 1e-9 * abs(measure[m].sum())


def test_oracle_group_join_vs_numpy():
    keys, measure = make_fact()
    dk, attrs = make_dim()
    t = oracle_table(keys, measure)
    t.set_dim(dk, attrs)
    plan = po.make_plan(aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
                        join=dict(dim=0, fact_col=0, group=True))
    rows = po.result_rows(t.query(plan))
    attr_of = dict(zip(dk.tolist(), [a.decode() for a in attrs]))
    exp = {}
    for k, v in zip(keys.tolist(), measure):
        a = attr_of.get(k)
        if a is None:
            continue
        s, c = exp.get(a, (0.0, 0))
        exp[a] = (s + v, c + 1)
    assert len(rows) == len(exp)
    for (gk,), vals in rows:
        s, c = exp[gk]
        assert vals[1] == float(c)
        assert abs(vals[0] - s) < 1e-9 * max(1.0, abs(s))


@pytest.mark.gpu
class TestJoinGpu:
    @pytest.fixture(scope="class")
    def setup(self):
        eng = se.Engine(device=0)
        keys, measure = make_fact(2_000_000, 50_000)
        dk, attrs = make_dim(50_000, 0.4, 6)
        t = eng.table_define("fact", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
        eng.ingest_columns(t, [{"data": keys}, {"data": measure}],
                           len(keys), batch_rows=300_000)
        dim = eng.dim_define("dim")
        eng.dim_put(dim, dk, attrs)

        ot = oracle_table(keys, measure)
        ot.set_dim(dk, attrs)
        yield eng, t, dim, ot
        eng.close()

    def test_semi_join(self, setup):
        eng, t, dim, ot = setup
        plan = abi.make_plan(table=t, aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
                             join=dict(dim=dim, fact_col=0))
        grows = eng.query(plan).rows()
        orows = po.result_rows(ot.query(po.make_plan(
            aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
            join=dict(dim=0, fact_col=0))))
        assert grows[0][1][1] == orows[0][1][1]
        assert abs(grows[0][1][0] - orows[0][1][0]) <= 1e-6 * abs(orows[0][1][0])

    def test_group_join(self, setup):
        eng, t, dim, ot = setup
        plan = abi.make_plan(table=t, aggs=[("sum", [(1, 0.0, 1.0)]),
                                            ("avg", [(1, 0.0, 1.0)]), ("count", [])],
                             join=dict(dim=dim, fact_col=0, group=True))
        grows = eng.query(plan).rows()
        orows = po.result_rows(ot.query(po.make_plan(
            aggs=[("sum", [(1, 0.0, 1.0)]), ("avg", [(1, 0.0, 1.0)]), ("count", [])],
            join=dict(dim=0, fact_col=0, group=True))))
        assert len(grows) == len(orows)
        for (gk, gv), (ok_, ov) in zip(grows, orows):
            assert gk == ok_
            assert gv[2] == ov[2]
            for a in (0, 1):
                assert abs(gv[a] - ov[a]) <= 1e-6 * max(1.0, abs(ov[a]))

    def test_group_join_with_filter_and_partials(self, setup):
        eng, t, dim, ot = setup
        mk = lambda mod: mod.make_plan(
            **(dict(table=t) if mod is abi else {}),
            preds=[dict(col=1, is_double=True, lo=25.0)],
            aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
            join=dict(dim=dim if mod is abi else 0, fact_col=0, group=True))
        q = eng.query(mk(abi))
        direct = q.rows()
        block = q.partials_host()
        q.merge_host(block, len(block), 1)
        assert q.rows() == direct
        orows = po.result_rows(ot.query(mk(po)))
        assert len(direct) == len(orows)
        for (gk, gv), (ok_, ov) in zip(direct, orows):
            assert gk == ok_ and gv[1] == ov[1]
            assert abs(gv[0] - ov[0]) <= 1e-6 * max(1.0, abs(ov[0]))


@pytest.mark.gpu
def test_group_join_25_nations(setup_factory=None):
    """TPC-H-realistic attr cardinality: 25 nations (> the 16-slot sweep
    kernel; routes through the LDS-accumulator paths)."""
    eng = se.Engine(device=0)
    keys, measure = make_fact(1_000_000, 20_000)
    dk = np.arange(20_000, dtype=np.int64)
    attrs = [b"NATION_%02d" % (int(k) % 25) for k in dk]
    t = eng.table_define("fact25", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": measure}],
                       len(keys), batch_rows=200_000)
    dim = eng.dim_define("nation25")
    eng.dim_put(dim, dk, attrs)
    plan = abi.make_plan(table=t, aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
                         join=dict(dim=dim, fact_col=0, group=True))
    rows = eng.query(plan).rows()
    attr_of = {int(k): a.decode() for k, a in zip(dk, attrs)}
    exp = {}
    for k, v in zip(keys.tolist(), measure):
        a = attr_of[k]
        s, c = exp.get(a, (0.0, 0))
        exp[a] = (s + v, c + 1)
    assert len(rows) == 25 == len(exp)
    for (gk,), vals in rows:
        s, c = exp[gk]
        assert vals[1] == float(c)
        assert abs(vals[0] - s) <= 1e-6 * max(1.0, abs(s))
    eng.close()


@pytest.mark.gpu
def test_incremental_dim_put_invalidates_tables():
    """a second dim_put must invalidate both the cached hash table AND the
    dense LUT (stale-LUT regression test)."""
    eng = se.Engine(device=0)
    keys, measure = make_fact(400_000, 5_000)
    t = eng.table_define("factinc", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": keys}, {"data": measure}],
                       len(keys), batch_rows=100_000)
    dim = eng.dim_define("dinc")
    dk1 = np.arange(0, 2_000, dtype=np.int64)
    eng.dim_put(dim, dk1, [b"A" for _ in dk1])
    plan = abi.make_plan(table=t, aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
                         join=dict(dim=dim, fact_col=0))
    r1 = eng.query(plan).rows()
    m1 = keys < 2_000
    assert r1[0][1][1] == float(m1.sum())
    # extend the dimension: results must now include the new keys
    dk2 = np.arange(2_000, 3_500, dtype=np.int64)
    eng.dim_put(dim, dk2, [b"B" for _ in dk2])
    r2 = eng.query(plan).rows()
    m2 = keys < 3_500
    assert r2[0][1][1] == float(m2.sum())
    assert abs(r2[0][1][0] - measure[m2].sum()) <= 1e-6 * abs(measure[m2].sum())
    eng.close()
