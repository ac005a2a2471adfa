"""Dictionary-pushdown string equality predicates: the engine resolves the
literal to its global dictionary id and filters on ids (ColumnTableScan's
dictionary filter pushdown); the oracle compares decoded bytes."""
import numpy as np
import pytest

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se


def test_oracle_string_eq_vs_numpy():
    n = 80_000
    rng = np.random.default_rng(81)
    keys = [b"CAT_%d" % v for v in rng.integers(0, 20, n)]
    vals = rng.random(n)
    t = po.OracleTable([po.T_STRING, po.T_DOUBLE])
    t.add_batch(n, [po.encode(po.T_STRING, po.ENC_DICT, keys),
                    po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, vals)])
    rows = po.result_rows(t.query(po.make_plan(
        preds=[dict(col=0, eq=b"CAT_7")],
        aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])))
    m = np.array([k == b"CAT_7" for k in keys])
    assert rows[0][1][1] == float(m.sum())
    assert abs(rows[0][1][0] - vals[m].sum()) <= 1e-9 * abs(vals[m].sum())


@pytest.mark.gpu
class TestStringPredGpu:
    def test_dict16_eq(self):
        n = 300_000
        rng = np.random.default_rng(82)
        keys = [b"CAT_%d" % v for v in rng.integers(0, 30, n)]
        vals = rng.random(n)
        eng = se.Engine(device=0)
        t = eng.table_define("tsp", [(abi.T_STRING, False), (abi.T_DOUBLE, False)])
        for bi, st in enumerate(range(0, n, 60_000)):
            en = min(n, st + 60_000)
            eng.batch_put(t, bi, bi, en - st,
                          [po.encode(po.T_STRING, po.ENC_DICT, keys[st:en]),
                           po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, vals[st:en])])
        q = eng.query(abi.make_plan(table=t, preds=[dict(col=0, eq=b"CAT_11")],
                                    aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])]))
        rows = q.rows()
        assert q.used_jit()
        m = np.array([k == b"CAT_11" for k in keys])
        assert rows[0][1][1] == float(m.sum())
        assert abs(rows[0][1][0] - vals[m].sum()) <= 1e-6 * abs(vals[m].sum())

        # absent literal -> zero rows, NULL sum
        q2 = eng.query(abi.make_plan(table=t, preds=[dict(col=0, eq=b"NOPE")],
                                     aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])]))
        r2 = q2.rows()
        assert r2[0][1][1] == 0.0 and r2[0][1][0] is None
        eng.close()

    def test_eq_combined_with_grouping_same_col(self):
        """predicate on the same dictionary column that keys the group-by:
        the premultiplied id must match the grouping layout."""
        n = 200_000
        rng = np.random.default_rng(83)
        k1 = [b"R%d" % v for v in rng.integers(0, 3, n)]
        k2 = [b"S%d" % v for v in rng.integers(0, 4, n)]
        vals = rng.random(n)
        eng = se.Engine(device=0)
        t = eng.table_define("tsg", [(abi.T_STRING, False), (abi.T_STRING, False),
                                     (abi.T_DOUBLE, False)])
        for bi, st in enumerate(range(0, n, 50_000)):
            en = min(n, st + 50_000)
            eng.batch_put(t, bi, bi, en - st,
                          [po.encode(po.T_STRING, po.ENC_DICT, k1[st:en]),
                           po.encode(po.T_STRING, po.ENC_DICT, k2[st:en]),
                           po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, vals[st:en])])
        plan = abi.make_plan(table=t, preds=[dict(col=0, eq=b"R1")],
                             group_cols=[0, 1],
                             aggs=[("sum", [(2, 0.0, 1.0)]), ("count", [])])
        rows = eng.query(plan).rows()
        exp = {}
        for a, b, v in zip(k1, k2, vals.tolist()):
            if a != b"R1":
                continue
            s, c = exp.get((a.decode(), b.decode()), (0.0, 0))
            exp[(a.decode(), b.decode())] = (s + v, c + 1)
        assert len(rows) == len(exp) == 4
        for gk, gv in rows:
            s, c = exp[(gk[0], gk[1])]
            assert gv[1] == float(c)
            assert abs(gv[0] - s) <= 1e-6 * max(1.0, abs(s))
        eng.close()

    def test_bigdict_eq_and_nullable(self):
        """BigDictionary (int32 index) + nullable column with nulls:
        general path, null rows never match."""
        n = 120_000
        rng = np.random.default_rng(84)
        card = 40_000   # > 32767: promotes to BigDictionary
        idx = rng.integers(0, card, n)
        keys = [b"K%05d" % v for v in idx]
        valid = (rng.random(n) >= 0.1).astype(np.uint8)
        keys = [k if ok else None for k, ok in zip(keys, valid)]
        vals = rng.random(n)
        eng = se.Engine(device=0)
        t = eng.table_define("tbd", [(abi.T_STRING, True), (abi.T_DOUBLE, False)])
        eng.batch_put(t, 0, 0, n,
                      [po.encode(po.T_STRING, po.ENC_BIGDICT, keys),
                       po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, vals)])
        target = b"K%05d" % int(idx[0])
        q = eng.query(abi.make_plan(table=t, preds=[dict(col=0, eq=target)],
                                    aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])]))
        rows = q.rows()
        m = np.array([k == target for k in keys])
        assert rows[0][1][1] == float(m.sum()) and m.sum() > 0
        assert abs(rows[0][1][0] - vals[m].sum()) <= 1e-6 * abs(vals[m].sum())
        eng.close()
