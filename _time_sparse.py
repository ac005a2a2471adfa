import time, numpy as np, os, sys
sys.path.insert(0, "/root/repo")
from snappydata_amd import abi, engine as se
n_batches, bat = 100, 600_000
rows = n_batches * bat
rng = np.random.default_rng(7)
keys = rng.integers(0, 1_000_000, rows).astype(np.int64)
keys = (keys * 2654435761) & ((1 << 53) - 1)   # spread like bench
vals = rng.random(rows)
e = se.Engine(device=0)
t = e.table_define("ts", [(abi.T_INT64, False), (abi.T_DOUBLE, False)])
e.ingest_columns(t, [{"data": keys}, {"data": vals}], rows, batch_rows=bat)
plan = abi.make_plan(table=t, group_cols=[0], aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
for mode in range(2):
    for it in range(3):
        t0 = time.perf_counter()
        q = e.query(plan)
        t1 = time.perf_counter()
        ng = q.num_groups()
        t2 = time.perf_counter()
        if it == 2:
            print(f"submit={1e3*(t1-t0):.2f}ms num_groups={1e3*(t2-t1):.2f}ms ngrp={ng} jit={q.used_jit()}")
        q.close() if hasattr(q, "close") else None
