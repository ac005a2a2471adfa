#!/usr/bin/env python3
"""End-to-end tour of the MI355X engine through the same surfaces a
SnappyData user knows — ingest encoded column batches, scan+filter+aggregate,
group-bys, broadcast joins, UPDATE/DELETE mutation, multi-shard merge.

Runs the host-side parts anywhere; the queries need one MI355X
(`python examples/demo.py` on a GPU box).
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from snappydata_amd import abi, engine as se  # noqa: E402


def main():
    eng = se.Engine(device=0)
    print("engine:", se.lib().sn_engine_arch().decode())

    # -- a lineitem-shaped table: CREATE TABLE ... USING column ------------
    t = eng.table_define("lineitem", [(abi.T_DOUBLE, False)] * 4 +
                         [(abi.T_STRING, False)] * 2 + [(abi.T_INT32, False)])
    eng.datagen_lineitem(t, 2_000_000, seed=1, batch_rows=200_000)
    print("rows:", eng.num_rows(t))

    # -- TPC-H Q6: conjunctive range filter + SUM --------------------------
    import datetime
    day = lambda y, m, d: (datetime.date(y, m, d) - datetime.date(1970, 1, 1)).days
    q6 = eng.query(abi.make_plan(
        table=t,
        preds=[dict(col=6, lo=day(1994, 1, 1), hi=day(1995, 1, 1), hi_strict=True),
               dict(col=2, is_double=True, lo=0.05, hi=0.07),
               dict(col=0, is_double=True, hi=24.0, hi_strict=True)],
        aggs=[("sum", [(1, 0.0, 1.0), (2, 0.0, 1.0)])]))
    print("Q6 revenue:", q6.rows()[0][1][0], "| query-compiled:", q6.used_jit(),
          "| kernel ms:", round(q6.kernel_ms(), 3))

    # -- Q1-style group-by over dictionary keys ----------------------------
    q1 = eng.query(abi.make_plan(
        table=t, group_cols=[4, 5],
        aggs=[("sum", [(0, 0.0, 1.0)]), ("avg", [(1, 0.0, 1.0)]), ("count", [])]))
    for key, vals in q1.rows():
        print("  group", key, "->", [round(v, 2) for v in vals[:2]], int(vals[2]))

    # -- dictionary-pushdown string equality -------------------------------
    qs = eng.query(abi.make_plan(
        table=t, preds=[dict(col=4, eq=b"A")],
        aggs=[("count", [])]))
    print("returnflag = 'A' rows:", int(qs.rows()[0][1][0]))

    # -- broadcast-dimension join ------------------------------------------
    dim = eng.dim_define("calendar")
    dk = np.arange(day(1994, 1, 1), day(1995, 1, 1), dtype=np.int64)
    eng.dim_put(dim, dk, [b"Y1994" for _ in dk])
    qj = eng.query(abi.make_plan(
        table=t, aggs=[("count", [])],
        join=dict(dim=dim, fact_col=6)))
    print("rows shipping in 1994:", int(qj.rows()[0][1][0]))

    # -- UPDATE then DELETE against a live batch ---------------------------
    tiny = eng.table_define("tiny", [(abi.T_DOUBLE, False)])
    base = np.arange(10, dtype=np.float64)
    from oracle import pyoracle as po
    eng.batch_put(tiny, 1, 0, 10, [po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, base)])
    upd = se.encode_update_delta(abi.T_DOUBLE, np.array([2], dtype=np.int32),
                                 10, np.array([100.0]))
    eng.batch_mutate(tiny, 1, 0, deltas=[(upd, None)])
    eng.batch_mutate(tiny, 1, 0, deltas=[(upd, None)],
                     delete_mask=se.encode_delete_mask(
                         np.array([9], dtype=np.int32), 10))
    qm = eng.query(abi.make_plan(table=tiny,
                                 aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])]))
    print("after UPDATE+DELETE:", qm.rows()[0][1])

    eng.close()


if __name__ == "__main__":
    main()
