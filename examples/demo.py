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

    # -- composite GROUP BY dim_attr, fact_col -----------------------------
    qjg = eng.query(abi.make_plan(
        table=t, group_cols=[4],                  # fact returnflag
        aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])],
        join=dict(dim=dim, fact_col=6, group=True)))
    print("1994 qty by (year, returnflag):")
    for key, vals in qjg.rows():
        print("  ", key, "->", round(vals[0], 2), int(vals[1]))

    # -- sparse group-by (open-address hash aggregate) + ORDER BY / TOP-K --
    rng = np.random.default_rng(7)
    sk = rng.integers(0, 50_000, 1_000_000).astype(np.int64) * (1 << 30)
    sv = rng.random(1_000_000)
    ts = eng.table_define("events", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
    eng.ingest_columns(ts, [{"data": sk}, {"data": sv}], len(sk),
                       batch_rows=250_000)
    qsp = eng.query(abi.make_plan(table=ts, group_cols=[0],
                                  aggs=[("sum", [(1, 0.0, 1.0)]),
                                        ("max", [(1, 0.0, 1.0)]),
                                        ("count", [])]))
    qsp.wait()
    total = qsp.num_groups()
    qsp.order_by(0, descending=True, k=3)   # TOP-3 groups by SUM
    print(f"sparse group-by: {total} groups; top-3 by sum:")
    for key, vals in qsp.rows():
        print("   key", key[0], "sum", round(vals[0], 2),
              "max", round(vals[1], 4), "rows", int(vals[2]))

    # -- colocated partitioned-partitioned join (build side on device) -----
    bt = eng.table_define("suppliers", [(abi.T_INT32, False), (abi.T_STRING, False)])
    bkeys = np.arange(0, 40_000, 2, dtype=np.int32)
    battrs = [b"N%d" % (int(k) % 4) for k in bkeys]
    eng.ingest_columns(bt, [{"data": bkeys},
                            {"data": b"".join(battrs),
                             "lens": np.array([len(a) for a in battrs],
                                              dtype=np.int32)}],
                       len(bkeys), batch_rows=10_000)
    fact = eng.table_define("orders", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    fk = rng.integers(0, 40_000, 500_000).astype(np.int32)
    fv = rng.random(500_000)
    eng.ingest_columns(fact, [{"data": fk}, {"data": fv}], len(fk),
                       batch_rows=125_000)
    d2 = eng.dim_define("suppliers_as_dim")
    eng.dim_from_table(d2, bt, key_col=0, attr_col=1)
    qc = eng.query(abi.make_plan(table=fact,
                                 aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])],
                                 join=dict(dim=d2, fact_col=0, group=True)))
    print("colocated join, orders by supplier nation:")
    for key, vals in qc.rows():
        print("  ", key[0], "->", round(vals[0], 2), int(vals[1]))

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
