"""Perf ablation driver (not a pytest): times the engine scan kernel across
plan shapes to isolate where Q6 time goes.  Run on a GPU box:
    python tests/perf_ablate.py
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from snappydata_amd import abi, engine as se
from tests import tpch_util as tu

def days(y,m,d):
    import datetime
    return (datetime.date(y,m,d)-datetime.date(1970,1,1)).days

eng = se.Engine(device=0)
t = eng.table_define("li", [(abi.T_DOUBLE, False)]*4 + [(abi.T_STRING, False)]*2 + [(abi.T_INT32, False)])
N = 60_000_000
eng.datagen_lineitem(t, N, seed=42, batch_rows=600_000)

Q, EP, DI, TX, RF, LS, SH = range(7)

shapes = {
    "1col_sum_nopred":      dict(preds=[], aggs=[("sum", [(Q,0.,1.)])]),
    "2col_sum_1pred":       dict(preds=[dict(col=SH, lo=days(1994,1,1))],
                                 aggs=[("sum", [(Q,0.,1.)])]),
    "4col_sum_nopred":      dict(preds=[], aggs=[("sum", [(Q,0.,1.)]), ("sum", [(EP,0.,1.)]),
                                                  ("sum", [(DI,0.,1.)]), ("sum", [(SH,0.,1.)])]),
    "q6_cols_sum_3pred":    dict(preds=[dict(col=SH, lo=days(1994,1,1), hi=days(1995,1,1), hi_strict=True),
                                        dict(col=DI, is_double=True, lo=0.05, hi=0.07),
                                        dict(col=Q, is_double=True, hi=24.0, hi_strict=True)],
                                 aggs=[("sum", [(EP,0.,1.), (DI,0.,1.)])]),
    "q6_1pred":             dict(preds=[dict(col=SH, lo=days(1994,1,1), hi=days(1995,1,1), hi_strict=True)],
                                 aggs=[("sum", [(EP,0.,1.), (DI,0.,1.)])]),
    "7col_sum_nopred":      dict(preds=[], aggs=[("sum", [(Q,0.,1.)]), ("sum", [(EP,0.,1.)]),
                                                  ("sum", [(DI,0.,1.)]), ("sum", [(TX,0.,1.)]),
                                                  ("sum", [(SH,0.,1.)])]),
    "grp_1agg":             dict(preds=[], group_cols=[RF, LS], aggs=[("sum", [(Q,0.,1.)])]),
    "grp_2agg":             dict(preds=[], group_cols=[RF, LS], aggs=[("sum", [(Q,0.,1.)]), ("count", [])]),
    "grp_4agg":             dict(preds=[], group_cols=[RF, LS],
                                 aggs=[("sum", [(Q,0.,1.)]), ("sum", [(EP,0.,1.)]),
                                       ("sum", [(DI,0.,1.)]), ("count", [])]),
    "grp_q1":               dict(preds=[dict(col=SH, hi=days(1997,10,2))], group_cols=[RF, LS],
                                 aggs=[("sum", [(Q,0.,1.)]), ("sum", [(EP,0.,1.)]),
                                       ("sum", [(EP,0.,1.), (DI,1.,-1.)]),
                                       ("sum", [(EP,0.,1.), (DI,1.,-1.), (TX,1.,1.)]),
                                       ("avg", [(Q,0.,1.)]), ("avg", [(EP,0.,1.)]),
                                       ("avg", [(DI,0.,1.)]), ("count", [])]),
    "grp_1agg_1grpcol":     dict(preds=[], group_cols=[RF], aggs=[("sum", [(Q,0.,1.)])]),
}

BYTES = {"1col_sum_nopred": 8, "2col_sum_1pred": 12, "4col_sum_nopred": 28,
         "q6_cols_sum_3pred": 28, "q6_1pred": 28, "7col_sum_nopred": 36,
         "grp_1agg": 12, "grp_2agg": 12, "grp_4agg": 28, "grp_q1": 40,
         "grp_1agg_1grpcol": 10}

for name, sh in shapes.items():
    print("running", name, flush=True); plan = abi.make_plan(table=t, **sh)
    for _ in range(3):
        q = eng.query(plan); q.wait(); km = q.kernel_ms(); q.close()
    kms = []
    for _ in range(8):
        q = eng.query(plan); q.wait(); kms.append(q.kernel_ms()); q.close()
    km = float(np.median(kms))
    gbs = N * BYTES[name] / (km/1e3) / 1e9
    print(f"{name:22s} kernel_ms={km:7.3f}  alg={BYTES[name]:2d}B/row  {gbs:7.0f} GB/s", flush=True)
