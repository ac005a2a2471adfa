"""Pin the CPU oracle against the reference's own TPC-H golden vectors.

Golden source: /root/reference/tests/common/src/main/resources/TPCH/RESULT/
Snappy_{1,6}.out computed from the bundled lineitem.tbl (30,201 rows),
validated by the reference's TPCHDUnitTest.scala:57-71,643-713.
These tests run in the build container where /root/reference exists; the
expectations are also committed in tpch_util.GOLDEN_* so parity survives on
boxes without the reference tree.
"""
import os

import pytest

from tests import tpch_util as tu
from oracle import pyoracle as po

HAVE_REF = os.path.isdir(tu.REFERENCE_TPCH)


@pytest.fixture(scope="module")
def lineitem():
    if not HAVE_REF:
        pytest.skip("reference TPCH data not present on this box")
    return tu.load_lineitem_tbl()


@pytest.fixture(scope="module")
def oracle_table(lineitem):
    t = po.OracleTable(tu.LINEITEM_DTYPES)
    for num_rows, cols, stats in tu.encode_lineitem_batches(lineitem, 4096):
        t.add_batch(num_rows, cols, stats=stats)
    return t


def test_golden_files_match_committed_copies():
    if not HAVE_REF:
        pytest.skip("reference tree absent")
    assert tu.load_golden(6) == tu.GOLDEN_Q6
    assert sorted(tu.load_golden(1)) == tu.GOLDEN_Q1


def test_q6_golden(oracle_table):
    res = oracle_table.query(tu.q6_plan())
    rows = po.result_rows(res)
    assert len(rows) == 1
    revenue = rows[0][1][0]
    assert tu.fmt(revenue) == tu.GOLDEN_Q6[0], revenue


def test_q1_golden(oracle_table):
    res = oracle_table.query(tu.q1_plan())
    rows = po.result_rows(res)
    assert len(rows) == 4
    lines = tu.q1_result_lines(rows)
    assert lines == tu.GOLDEN_Q1


def test_q1_multithreaded_matches_golden(oracle_table):
    """The OpenMP CPU-baseline path must produce the same printed results
    (different summation order; equality at the golden 4-decimal precision)."""
    res = oracle_table.query(tu.q1_plan(), nthreads=4)
    rows = po.result_rows(res)
    assert tu.q1_result_lines(rows) == tu.GOLDEN_Q1


def test_q6_stats_skip(lineitem, oracle_table):
    """Q6's shipdate range must skip batches whose stats exclude 1994, without
    changing the result.  Batches are clustered by shipdate so per-batch
    min/max ranges actually exclude the predicate window (mirrors the
    reference's per-batch ColumnStatsSchema skip, ColumnTableScan.scala:820-963)."""
    import numpy as np
    order = np.argsort(lineitem["ship"], kind="stable")
    clustered = {
        "qty": lineitem["qty"][order], "ep": lineitem["ep"][order],
        "disc": lineitem["disc"][order], "tax": lineitem["tax"][order],
        "rf": [lineitem["rf"][i] for i in order],
        "ls": [lineitem["ls"][i] for i in order],
        "ship": lineitem["ship"][order],
    }
    t = po.OracleTable(tu.LINEITEM_DTYPES)
    for num_rows, cols, stats in tu.encode_lineitem_batches(clustered, 4096):
        t.add_batch(num_rows, cols, stats=stats)
    res = t.query(tu.q6_plan())
    assert res.batches_seen == (len(lineitem["qty"]) + 4095) // 4096
    assert 0 < res.batches_skipped < res.batches_seen
    # skipping must not change the result
    assert tu.fmt(po.result_rows(res)[0][1][0]) == tu.GOLDEN_Q6[0]
    # unclustered table sees every batch and agrees too
    r2 = oracle_table.query(tu.q6_plan())
    assert tu.fmt(po.result_rows(r2)[0][1][0]) == tu.GOLDEN_Q6[0]


def test_q6_single_batch_same_result(lineitem):
    t = po.OracleTable(tu.LINEITEM_DTYPES)
    for num_rows, cols, stats in tu.encode_lineitem_batches(lineitem, 10**9):
        t.add_batch(num_rows, cols, stats=stats)
    rows = po.result_rows(t.query(tu.q6_plan()))
    assert tu.fmt(rows[0][1][0]) == tu.GOLDEN_Q6[0]
