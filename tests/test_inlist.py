"""IN-list predicates (the reference's Q12/Q19-class `col IN (...)`
filters, compiled into the generated scan loop): dictionary strings
resolve to premultiplied dictionary ids at submit; integer lists build a
bitmap LUT when the value span is dense (query-compiled) or a sorted
binary-searched list when wide (interpreted kernels)."""
import numpy as np
import pytest

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se

REL = 1e-9


@pytest.fixture
def eng():
    e = se.Engine(device=0)
    yield e
    e.close()


def test_oracle_in_list_cpu():
    n = 200_000
    rng = np.random.default_rng(157)
    modes = [b"AIR", b"SHIP", b"RAIL", b"TRUCK", b"MAIL"]
    m = [modes[v] for v in rng.integers(0, 5, n)]
    qty = rng.integers(1, 50, n).astype(np.int32)
    t = po.OracleTable([po.T_STRING, po.T_INT32])
    t.add_batch(n, [po.encode(po.T_STRING, po.ENC_DICT, m),
                    po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, qty)])
    rows = po.result_rows(t.query(po.make_plan(
        preds=[dict(col=0, **{"in": [b"AIR", b"RAIL"]}),
               dict(col=1, **{"in": [3, 7, 11]})],
        aggs=[("count", [])])))
    mask = np.array([x in (b"AIR", b"RAIL") for x in m]) & np.isin(qty, [3, 7, 11])
    assert rows[0][1][0] == float(mask.sum()) and mask.sum() > 0


@pytest.mark.gpu
def test_in_list_dict_strings_jit(eng):
    """Q12 shape: l_shipmode IN ('AIR','RAIL') + range pred; the dictionary
    IN resolves to a gid bitmap and stays on the query-compiled path."""
    n = 600_000
    rng = np.random.default_rng(163)
    modes = [b"AIR", b"SHIP", b"RAIL", b"TRUCK", b"MAIL"]
    m = [modes[v] for v in rng.integers(0, 5, n)]
    qty = rng.random(n) * 50
    cols = [po.encode(po.T_STRING, po.ENC_DICT, m),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, qty)]
    t = eng.table_define("tq12", [(abi.T_STRING, False), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 0, 0, n, cols)
    plan_kw = dict(preds=[dict(col=0, **{"in": [b"AIR", b"RAIL"]}),
                          dict(col=1, is_double=True, hi=25.0, hi_strict=True)],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    q = eng.query(abi.make_plan(table=t, **plan_kw))
    grows = q.rows()
    assert q.used_jit()
    ot = po.OracleTable([po.T_STRING, po.T_DOUBLE])
    ot.add_batch(n, cols)
    orows = po.result_rows(ot.query(po.make_plan(**plan_kw)))
    assert grows[0][1][1] == orows[0][1][1]
    assert abs(grows[0][1][0] - orows[0][1][0]) <= REL * max(1.0, orows[0][1][0])
    mask = np.array([x in (b"AIR", b"RAIL") for x in m]) & (qty < 25.0)
    assert grows[0][1][1] == float(mask.sum())


@pytest.mark.gpu
def test_in_list_grouped_by_mode(eng):
    """IN filter + GROUP BY the same dictionary column (premultiplied ids
    must agree between the filter bitmap and the group slots)."""
    n = 400_000
    rng = np.random.default_rng(167)
    modes = [b"AIR", b"SHIP", b"RAIL", b"TRUCK"]
    m = [modes[v] for v in rng.integers(0, 4, n)]
    w = rng.random(n)
    cols = [po.encode(po.T_STRING, po.ENC_DICT, m),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    t = eng.table_define("ting", [(abi.T_STRING, False), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 0, 0, n, cols)
    plan_kw = dict(preds=[dict(col=0, **{"in": [b"AIR", b"TRUCK"]})],
                   group_cols=[0],
                   aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    grows = eng.query(abi.make_plan(table=t, **plan_kw)).rows()
    ot = po.OracleTable([po.T_STRING, po.T_DOUBLE])
    ot.add_batch(n, cols)
    orows = po.result_rows(ot.query(po.make_plan(**plan_kw)))
    assert [k for k, _ in grows] == [("AIR",), ("TRUCK",)]
    assert len(grows) == len(orows)
    for (gk, gv), (ok_, ov) in zip(grows, orows):
        assert gk == ok_ and gv[1] == ov[1]
        assert abs(gv[0] - ov[0]) <= 1e-9 * max(1.0, abs(ov[0]))


@pytest.mark.gpu
def test_in_list_wide_int64_span_interpreted(eng):
    """int64 IN over a span too wide for a bitmap: sorted-list binary
    search on the interpreted kernels (exact beyond 2^53)."""
    n = 300_000
    rng = np.random.default_rng(173)
    universe = (np.arange(50, dtype=np.int64) * (1 << 40)) + (1 << 57)
    v = universe[rng.integers(0, 50, n)]
    w = rng.random(n)
    t = eng.table_define("tin64", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": v}, {"data": w}], n, batch_rows=100_000)
    picks = [int(universe[3]), int(universe[17]), int(universe[41])]
    q = eng.query(abi.make_plan(
        table=t, preds=[dict(col=0, **{"in": picks})],
        aggs=[("count", [])]))
    cnt = q.rows()[0][1][0]
    assert not q.used_jit()        # list form stays interpreted
    assert cnt == float(np.isin(v, picks).sum()) > 0


@pytest.mark.gpu
def test_in_list_empty_and_absent_literals(eng):
    n = 50_000
    rng = np.random.default_rng(179)
    m = [b"A" if x else b"B" for x in rng.integers(0, 2, n)]
    w = np.ones(n)
    cols = [po.encode(po.T_STRING, po.ENC_DICT, m),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)]
    t = eng.table_define("tinempty", [(abi.T_STRING, False), (abi.T_DOUBLE, False)])
    eng.batch_put(t, 0, 0, n, cols)
    # literals absent from the dictionary -> zero rows
    q = eng.query(abi.make_plan(table=t,
                                preds=[dict(col=0, **{"in": [b"ZZZ", b"YYY"]})],
                                aggs=[("count", [])]))
    assert q.rows()[0][1][0] == 0.0
    # mix of present and absent literals
    q2 = eng.query(abi.make_plan(table=t,
                                 preds=[dict(col=0, **{"in": [b"A", b"NOPE"]})],
                                 aggs=[("count", [])]))
    assert q2.rows()[0][1][0] == float(sum(1 for x in m if x == b"A"))
