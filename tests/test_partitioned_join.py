"""Partitioned-partitioned (colocated) join — SURVEY §8(f)3 scoped to the
reference's colocated case: both tables partition on the join key, so each
shard joins bucket-locally with NO exchange (GemFire colocation).  The
build side is a COLUMN TABLE whose (key, attr) rows populate the probe
table ON DEVICE (sn_dim_from_table — the HashJoinExec per-task
ObjectHashSet build with HashedObjectCache reuse, HashJoinExec.scala:
285-520, :449-470); the probe then runs the existing join kernels.

Oracle leg: the same build rows fed to the oracle's dimension path
(semantically identical: inner join on a unique key)."""
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


def _mkfact(eng, keys, ep, batch=100_000, name="fact"):
    t = eng.table_define(name, [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    n = len(keys)
    for st in range(0, n, batch):
        en = min(n, st + batch)
        eng.ingest_columns(t, [{"data": keys[st:en]}, {"data": ep[st:en]}],
                           en - st, batch_rows=batch, first_bucket=st // batch)
    return t


@pytest.mark.gpu
def test_colocated_join_group_by_attr(eng):
    n = 1_000_000
    rng = np.random.default_rng(113)
    keyspace = 200_000
    fkeys = rng.integers(0, keyspace, n).astype(np.int32)
    ep = rng.random(n) * 1e4
    fact = _mkfact(eng, fkeys, ep)

    # build side: 40% of the keyspace, 8 distinct nation attrs
    bkeys = np.sort(rng.choice(keyspace, size=keyspace * 2 // 5,
                               replace=False)).astype(np.int32)
    battrs = [b"NATION_%d" % (int(k) % 8) for k in bkeys]
    blens = np.ones(len(bkeys), dtype=np.int32) * 8  # "NATION_x" is 8 bytes
    bt = eng.table_define("build", [(abi.T_INT32, False), (abi.T_STRING, False)])
    eng.ingest_columns(bt, [{"data": bkeys},
                            {"data": b"".join(battrs), "lens": blens}],
                       len(bkeys), batch_rows=60_000)

    dim = eng.dim_define("from_build")
    eng.dim_from_table(dim, bt, key_col=0, attr_col=1)
    plan = abi.make_plan(table=fact,
                         aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
                         join=dict(dim=dim, fact_col=0, group=True))
    grows = eng.query(plan).rows()

    ot = po.OracleTable([po.T_INT32, po.T_DOUBLE])
    ot.add_batch(n, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, fkeys),
                     po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, ep)])
    ot.set_dim(bkeys.astype(np.int64), battrs)
    orows = po.result_rows(ot.query(po.make_plan(
        aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
        join=dict(dim=0, fact_col=0, group=True)), nthreads=16))
    assert len(grows) == len(orows) == 8
    for (gk, gv), (ok_, ov) in zip(grows, orows):
        assert gk == ok_
        assert gv[1] == ov[1]
        assert abs(gv[0] - ov[0]) <= REL * max(1.0, abs(ov[0]))


@pytest.mark.gpu
def test_colocated_semi_join(eng):
    n = 400_000
    rng = np.random.default_rng(127)
    fkeys = rng.integers(0, 50_000, n).astype(np.int32)
    ep = rng.random(n)
    fact = _mkfact(eng, fkeys, ep, name="fact2")
    bkeys = np.unique(rng.integers(0, 50_000, 18_000)).astype(np.int32)
    bt = eng.table_define("build2", [(abi.T_INT32, False)])
    eng.ingest_columns(bt, [{"data": bkeys}], len(bkeys), batch_rows=7_000)
    dim = eng.dim_define("semi_from_build")
    eng.dim_from_table(dim, bt, key_col=0)
    q = eng.query(abi.make_plan(table=fact, aggs=[("count", [])],
                                join=dict(dim=dim, fact_col=0)))
    cnt = q.rows()[0][1][0]
    assert cnt == float(np.isin(fkeys, bkeys).sum())


@pytest.mark.gpu
def test_colocated_join_sharded_local(eng):
    """Colocation semantics: both sides bucketed by hash(key)%nbuckets;
    two shard engines join locally with no exchange, and the merged counts
    equal the unsharded join."""
    n = 300_000
    rng = np.random.default_rng(131)
    keyspace = 30_000
    fkeys = rng.integers(0, keyspace, n).astype(np.int32)
    ep = rng.random(n)
    bkeys = np.unique(rng.integers(0, keyspace, 12_000)).astype(np.int32)

    def load(e2):
        # colocate: bucket = key % 16 for BOTH sides (both tables partition
        # on the join key, the reference's colocation requirement)
        fact = e2.table_define("f", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
        bt = e2.table_define("b", [(abi.T_INT32, False)])
        for bkt in range(16):
            fm = fkeys % 16 == bkt
            if fm.sum():
                e2.ingest_columns(fact, [{"data": fkeys[fm]}, {"data": ep[fm]}],
                                  int(fm.sum()), batch_rows=1 << 20,
                                  first_bucket=bkt)
            bm = bkeys % 16 == bkt
            if bm.sum():
                e2.ingest_columns(bt, [{"data": bkeys[bm]}], int(bm.sum()),
                                  batch_rows=1 << 20, first_bucket=bkt)
        dim = e2.dim_define("d")
        e2.dim_from_table(dim, bt, key_col=0)
        return fact, dim

    fact, dim = load(eng)
    ref = eng.query(abi.make_plan(table=fact, aggs=[("count", [])],
                                  join=dict(dim=dim, fact_col=0))).rows()[0][1][0]

    e0 = se.Engine(device=0, shard_rank=0, shard_count=2)
    e1 = se.Engine(device=0, shard_rank=1, shard_count=2)
    try:
        total = 0.0
        for e2 in (e0, e1):
            f2, d2 = load(e2)
            total += e2.query(abi.make_plan(
                table=f2, aggs=[("count", [])],
                join=dict(dim=d2, fact_col=0))).rows()[0][1][0]
        assert total == ref == float(np.isin(fkeys, bkeys).sum())
    finally:
        e0.close()
        e1.close()


@pytest.mark.gpu
def test_join_build_rejects_conflicting_duplicates(eng):
    keys = np.array([1, 2, 3, 2], dtype=np.int32)
    attrs = [b"A", b"B", b"C", b"D"]     # key 2 -> B and D: conflict
    lens = np.array([1, 1, 1, 1], dtype=np.int32)
    bt = eng.table_define("dup", [(abi.T_INT32, False), (abi.T_STRING, False)])
    eng.ingest_columns(bt, [{"data": keys},
                            {"data": b"".join(attrs), "lens": lens}], 4,
                       batch_rows=4)
    dim = eng.dim_define("dupdim")
    with pytest.raises(se.EngineError):
        eng.dim_from_table(dim, bt, key_col=0, attr_col=1)
