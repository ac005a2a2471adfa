"""Bench-scale parity gate (GPU): Q1 + Q6 at SF=10 scale (60M rows, the
bench generator's own data) — the perf-headline workload itself is
GPU-vs-oracle parity-checked, not only the small fixture tables.

Double sums at 600M-row magnitudes stress the 1e-6 relative budget far more
than the 1M-row tests; this gate runs the exact generator bench.py uses
(sn_gen_lineitem_arrays / sn_datagen_lineitem share one seeded per-row
function, so engine and oracle see identical bytes).
"""
import os

import numpy as np
import pytest

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se
from tests import tpch_util as tu

pytestmark = pytest.mark.gpu

ROWS = 60_000_000
BATCH = 600_000
SEED = 42
REL = 1e-6


@pytest.fixture(scope="module")
def tables():
    eng = se.Engine(device=0)
    t = eng.table_define("li_sf10", [(abi.T_DOUBLE, False)] * 4 +
                         [(abi.T_STRING, False)] * 2 + [(abi.T_INT32, False)])
    put = eng.datagen_lineitem(t, ROWS, seed=SEED, batch_rows=BATCH)
    assert put == ROWS
    ot = po.OracleTable(tu.LINEITEM_DTYPES)
    for start in range(0, ROWS, BATCH):
        n = min(BATCH, ROWS - start)
        d = se.gen_lineitem_arrays(start, n, SEED)
        for num_rows, cols, stats in tu.encode_lineitem_batches(d, n):
            ot.add_batch(num_rows, cols, stats=stats)
    yield eng, t, ot
    eng.close()


def _compare(grows, orows, count_aggs):
    assert len(grows) == len(orows)
    for (gk, gv), (ok_, ov) in zip(grows, orows):
        assert gk == ok_
        for a, (g, o) in enumerate(zip(gv, ov)):
            if a in count_aggs:
                assert g == o, (a, g, o)
            elif o is None:
                assert g is None
            else:
                assert abs(g - o) <= REL * max(1.0, abs(o)), (a, g, o)


def test_q6_sf10_scale_parity(tables):
    eng, t, ot = tables
    q = eng.query(abi.make_plan(
        table=t,
        preds=[dict(col=tu.COL_SHIP, lo=tu.days(1994, 1, 1),
                    hi=tu.days(1995, 1, 1), hi_strict=True),
               dict(col=tu.COL_DISC, is_double=True, lo=0.05, hi=0.07),
               dict(col=tu.COL_QTY, is_double=True, hi=24.0, hi_strict=True)],
        aggs=[("sum", [(tu.COL_EP, 0.0, 1.0), (tu.COL_DISC, 0.0, 1.0)]),
              ("count", [])]))
    grows = q.rows()
    assert q.used_jit()
    nthreads = min(os.cpu_count() or 1, 128)
    ores = ot.query(po.make_plan(
        preds=[dict(col=tu.COL_SHIP, lo=tu.days(1994, 1, 1),
                    hi=tu.days(1995, 1, 1), hi_strict=True),
               dict(col=tu.COL_DISC, is_double=True, lo=0.05, hi=0.07),
               dict(col=tu.COL_QTY, is_double=True, hi=24.0, hi_strict=True)],
        aggs=[("sum", [(tu.COL_EP, 0.0, 1.0), (tu.COL_DISC, 0.0, 1.0)]),
              ("count", [])]), nthreads=nthreads)
    _compare(grows, po.result_rows(ores), count_aggs={1})


def test_q1_sf10_scale_parity(tables):
    eng, t, ot = tables
    cutoff = tu.days(1997, 12, 31) - 90
    aggs = [("sum", [(tu.COL_QTY, 0.0, 1.0)]),
            ("sum", [(tu.COL_EP, 0.0, 1.0)]),
            ("sum", [(tu.COL_EP, 0.0, 1.0), (tu.COL_DISC, 1.0, -1.0)]),
            ("sum", [(tu.COL_EP, 0.0, 1.0), (tu.COL_DISC, 1.0, -1.0),
                     (tu.COL_TAX, 1.0, 1.0)]),
            ("avg", [(tu.COL_QTY, 0.0, 1.0)]),
            ("avg", [(tu.COL_EP, 0.0, 1.0)]),
            ("avg", [(tu.COL_DISC, 0.0, 1.0)]),
            ("count", [])]
    q = eng.query(abi.make_plan(
        table=t, preds=[dict(col=tu.COL_SHIP, hi=cutoff)],
        group_cols=[tu.COL_RF, tu.COL_LS], aggs=aggs))
    grows = q.rows()
    assert q.used_jit()
    assert len(grows) == 6
    nthreads = min(os.cpu_count() or 1, 128)
    ores = ot.query(po.make_plan(
        preds=[dict(col=tu.COL_SHIP, hi=cutoff)],
        group_cols=[tu.COL_RF, tu.COL_LS], aggs=aggs), nthreads=nthreads)
    _compare(grows, po.result_rows(ores), count_aggs={7})


def test_rows_scanned_consistency(tables):
    """SQLMetrics parity at scale: both sides saw every row."""
    eng, t, ot = tables
    q = eng.query(abi.make_plan(table=t, aggs=[("count", [])]))
    rows = q.rows()
    assert rows[0][1][0] == float(ROWS)
    res = q.result()
    assert res.rows_scanned == ROWS
    assert res.batches_seen == ROWS // BATCH
