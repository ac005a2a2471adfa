"""CPU-side guards for the driver-facing bench contract: plan builders,
traffic calibration lookup, and the CPU-baseline leg (the GPU measurement
itself needs a device and is exercised by the driver)."""
import json
import os

import numpy as np

import bench
from snappydata_amd import abi


def test_workload_table_shape():
    for name, (plan_fn, rows, bpr) in bench.WORKLOADS.items():
        assert rows > 0 and bpr > 0
        assert name in ("star_join_sf10", "config1_sum_where",
                        "sparse_group_sf10") or plan_fn


def test_q6_q1_plan_construction():
    q6 = bench.q6_plan(3)
    assert q6.table == 3 and q6.npreds == 3 and q6.naggs == 1
    assert q6.ngroup == 0
    q1 = bench.q1_plan(0)
    assert q1.npreds == 1 and q1.ngroup == 2 and q1.naggs == 8
    # DECIMAL-folded Q6 discount bounds (golden-pinned)
    disc = [q6.preds[i] for i in range(3) if q6.preds[i].col == bench.COL_DISC][0]
    assert disc.lo_d == 0.05 and disc.hi_d == 0.07


def test_pmc_traffic_lookup():
    v = bench._pmc_traffic("tpch_q6_lineitem_sf10", 60_000_000)
    assert v is not None and 1.5e9 < v < 2.0e9
    assert bench._pmc_traffic("nonexistent_workload", 1) is None


def test_traffic_json_committed_and_sane():
    cal = json.load(open(os.path.join(bench.REPO, "profiles", "traffic.json")))
    assert 27 < cal["tpch_q6_lineitem_sf10"]["bytes_per_row"] < 30
    assert 39 < cal["tpch_q1_lineitem_sf10"]["bytes_per_row"] < 42


def test_cpu_baseline_leg_contract():
    r = bench.cpu_baseline_leg("tpch_q6_lineitem_sf10", 42, target_seconds=0.3,
                               sample_rows=2_400_000)
    assert r["unit"] == "rows/s" and r["kind"] == "port"
    assert r["value"] > 1e6 and r["cores"] >= 1
    assert "oracle" in r["sample"]
