"""INT64 predicate exactness beyond 2^53: the engine keeps int64 columns as
raw bits in the canonical image and compares as integers (i64_mask), so
bounds that round to the SAME double must still filter exactly."""
import numpy as np
import pytest

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se

BASE = 1 << 60   # doubles round to multiples of 256 here


def make_data(n=100_000, seed=61):
    rng = np.random.default_rng(seed)
    vals = BASE + rng.integers(0, 50, n).astype(np.int64)
    w = rng.random(n)
    return vals, w


def the_plan(mod, table=None):
    kw = dict(table=table) if table is not None else {}
    return mod.make_plan(**kw,
                         preds=[dict(col=0, lo=BASE + 5, hi=BASE + 7)],
                         aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])


def test_oracle_int64_exact_bounds():
    vals, w = make_data()
    t = po.OracleTable([po.T_INT64, po.T_DOUBLE])
    t.add_batch(len(vals), [po.encode(po.T_INT64, po.ENC_UNCOMPRESSED, vals),
                            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, w)])
    rows = po.result_rows(t.query(the_plan(po)))
    m = (vals >= BASE + 5) & (vals <= BASE + 7)
    assert rows[0][1][1] == float(m.sum()) and m.sum() > 0
    # double rounding would match far more rows (BASE..BASE+255 collapse)
    assert float(np.float64(BASE + 5)) == float(np.float64(BASE + 7))


@pytest.mark.gpu
def test_engine_int64_exact_bounds():
    vals, w = make_data()
    eng = se.Engine(device=0)
    t = eng.table_define("t64", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": vals}, {"data": w}], len(vals),
                       batch_rows=25_000)
    q = eng.query(the_plan(abi, table=t))
    rows = q.rows()
    assert q.used_jit()
    m = (vals >= BASE + 5) & (vals <= BASE + 7)
    assert rows[0][1][1] == float(m.sum())
    exp = w[m].sum()
    assert abs(rows[0][1][0] - exp) <= 1e-9 * max(1.0, exp)
    eng.close()
