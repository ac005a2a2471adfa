"""Byte-level layout pins: re-encode the committed fixture inputs and demand
byte equality with the committed blobs; decode the committed blobs and demand
value equality.  These vectors are this build's byte-level pin for the column
formats (the reference has no golden bytes — SURVEY.md §8(c)).

Also replays TPC-H Q1/Q6 from the committed lineitem batch fixture, which is
how GPU-box runs (no /root/reference) reproduce Snappy_1.out/Snappy_6.out.
"""
import json
import os

import numpy as np
import pytest

from oracle import pyoracle as po
from tests import tpch_util as tu
from tests.golden.make_fixtures import read_batches

HERE = os.path.dirname(os.path.abspath(__file__))
GOLD = os.path.join(HERE, "golden")


def load_vectors():
    z = np.load(os.path.join(GOLD, "vectors.npz"))
    manifest = json.loads(bytes(z["__manifest"]).decode())
    return z, manifest


def unpack_strings(lens, payload):
    vals, off = [], 0
    raw = bytes(payload)
    for L in lens:
        if L < 0:
            vals.append(None)
        else:
            vals.append(raw[off:off + L])
            off += L
    return vals


def test_reencode_matches_committed_blobs():
    z, manifest = load_vectors()
    for m in manifest:
        name, dtype, enc, count = m["name"], m["dtype"], m["enc"], m["count"]
        valid = z[f"{name}__valid"] if f"{name}__valid" in z else None
        if dtype == po.T_STRING:
            values = unpack_strings(z[f"{name}__len"], z[f"{name}__payload"])
        else:
            values = z[f"{name}__values"]
        blob = po.encode(dtype, enc, values, valid)
        assert blob == bytes(z[f"{name}__blob"]), f"layout drift in {name}"


def test_decode_committed_blobs():
    z, manifest = load_vectors()
    for m in manifest:
        name, dtype, count = m["name"], m["dtype"], m["count"]
        blob = bytes(z[f"{name}__blob"])
        got, gvalid = po.decode(dtype, blob, count)
        valid = z[f"{name}__valid"] if f"{name}__valid" in z else None
        if dtype == po.T_STRING:
            exp = unpack_strings(z[f"{name}__len"], z[f"{name}__payload"])
            assert got == exp, name
        else:
            exp = z[f"{name}__values"]
            m_ = np.ones(count, bool) if valid is None else valid.astype(bool)
            assert np.array_equal(np.asarray(got)[m_], exp[m_]), name


@pytest.fixture(scope="module")
def fixture_table():
    t = po.OracleTable(tu.LINEITEM_DTYPES)
    for num_rows, cols, stats in read_batches(os.path.join(GOLD, "lineitem_batches.bin")):
        t.add_batch(num_rows, cols, stats=stats)
    return t


def test_q6_from_fixture(fixture_table):
    rows = po.result_rows(fixture_table.query(tu.q6_plan()))
    assert tu.fmt(rows[0][1][0]) == tu.GOLDEN_Q6[0]


def test_q1_from_fixture(fixture_table):
    rows = po.result_rows(fixture_table.query(tu.q1_plan()))
    assert tu.q1_result_lines(rows) == tu.GOLDEN_Q1
