"""Property-based encode/decode round trips (hypothesis): the oracle's
decoders against its encoders and the PRODUCT builder's encoders, across
dtypes, null patterns, encodings and sizes — the reference only has
round-trip tests for encoders (ColumnEncodersTest.scala:26-33), so breadth
here is the byte-layout safety net."""
import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se

DTYPES = [
    (po.T_DOUBLE, np.float64),
    (po.T_INT32, np.int32),
    (po.T_INT64, np.int64),
    (po.T_INT16, np.int16),
    (po.T_FLOAT, np.float32),
]


@settings(max_examples=30, deadline=None)
@given(st.integers(0, 4), st.integers(1, 3000), st.integers(0, 2**31 - 1),
       st.floats(0.0, 0.9))
def test_uncompressed_nullable_roundtrip(di, n, seed, null_frac):
    dt, npdt = DTYPES[di]
    rng = np.random.default_rng(seed)
    if npdt in (np.float64, np.float32):
        vals = rng.random(n).astype(npdt)
    else:
        info = np.iinfo(npdt)
        vals = rng.integers(info.min, info.max, n).astype(npdt)
    valid = (rng.random(n) >= null_frac).astype(np.uint8)
    blob = po.encode(dt, po.ENC_UNCOMPRESSED, vals, valid=valid)
    out_vals, out_valid = po.decode(dt, blob, n)
    np.testing.assert_array_equal(out_valid, valid)
    np.testing.assert_array_equal(out_vals[valid.astype(bool)],
                                  vals[valid.astype(bool)])


@settings(max_examples=20, deadline=None)
@given(st.integers(1, 2000), st.integers(0, 2**31 - 1), st.integers(1, 40))
def test_dictionary_string_roundtrip(n, seed, card):
    rng = np.random.default_rng(seed)
    keys = [b"K%05d" % v for v in rng.integers(0, card, n)]
    blob = po.encode(po.T_STRING, po.ENC_DICT, keys)
    out, valid = po.decode(po.T_STRING, blob, n)
    assert list(out) == keys
    assert valid.all()


@settings(max_examples=20, deadline=None)
@given(st.integers(1, 1500), st.integers(0, 2**31 - 1))
def test_builder_matches_oracle_bytes(n, seed):
    """the PRODUCT encoder must emit byte-identical blobs to the oracle's
    independent restatement"""
    rng = np.random.default_rng(seed)
    f64 = rng.random(n)
    i32 = rng.integers(-1000, 1000, n).astype(np.int32)
    assert se.encode_column(abi.T_DOUBLE, f64) == \
        po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, f64)
    assert se.encode_column(abi.T_INT32, i32) == \
        po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, i32)


@settings(max_examples=15, deadline=None)
@given(st.integers(1, 1000), st.integers(0, 2**31 - 1), st.integers(1, 50),
       st.integers(0, 30))
def test_delete_delta_oracle_vs_numpy(n, seed, ndel, nupd):
    """delete mask + update delta agreement with a direct numpy model"""
    rng = np.random.default_rng(seed)
    f64 = rng.random(n)
    dels = np.unique(rng.integers(0, n, min(ndel, n))).astype(np.int32)
    dmask = po.encode_delete(dels, n)
    deltas = [(None, None)]
    upd = np.unique(rng.integers(0, n, nupd)).astype(np.int32) if nupd else None
    newv = None
    if upd is not None and len(upd):
        newv = rng.random(len(upd)) * 5
        deltas = [(po.encode_delta(po.T_DOUBLE, po.ENC_UNCOMPRESSED,
                                   upd, n, newv), None)]
    t = po.OracleTable([po.T_DOUBLE])
    t.add_batch(-n if deltas[0][0] else n,
                [po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, f64)],
                delete_mask=dmask, deltas=deltas)
    rows = po.result_rows(t.query(po.make_plan(
        aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])])))
    ref = f64.copy()
    if newv is not None:
        ref[upd] = newv
    keep = np.ones(n, bool)
    keep[dels] = False
    assert rows[0][1][1] == float(keep.sum())
    exp = ref[keep].sum()
    if keep.sum() == 0:
        assert rows[0][1][0] is None
    else:
        assert abs(rows[0][1][0] - exp) <= 1e-9 * max(1.0, abs(exp))


def test_batch_order_invariance():
    """scan+agg results must not depend on batch insertion order."""
    n = 120_000
    rng = np.random.default_rng(67)
    i32 = rng.integers(0, 500, n).astype(np.int32)
    f64 = rng.random(n)
    plan = po.make_plan(preds=[dict(col=0, hi=250, hi_strict=True)],
                        aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])
    chunks = [(s, min(n, s + 30_000)) for s in range(0, n, 30_000)]

    def run(order):
        t = po.OracleTable([po.T_INT32, po.T_DOUBLE])
        for s, e in order:
            t.add_batch(e - s, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, i32[s:e]),
                                po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, f64[s:e])])
        return po.result_rows(t.query(plan))

    a = run(chunks)
    b = run(list(reversed(chunks)))
    assert a[0][1][1] == b[0][1][1]
    assert abs(a[0][1][0] - b[0][1][0]) <= 1e-9 * max(1.0, abs(a[0][1][0]))


def test_linearity_disjoint_union():
    """SUM/COUNT over a union of disjoint row sets equals the sum of the
    parts (the partial->final merge law at the semantic level)."""
    n = 100_000
    rng = np.random.default_rng(68)
    f64 = rng.random(n)
    plan = po.make_plan(aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])])

    def run(lo, hi):
        t = po.OracleTable([po.T_DOUBLE])
        t.add_batch(hi - lo, [po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, f64[lo:hi])])
        return po.result_rows(t.query(plan))[0][1]

    whole = run(0, n)
    left = run(0, n // 3)
    right = run(n // 3, n)
    assert whole[1] == left[1] + right[1] == float(n)
    assert abs(whole[0] - (left[0] + right[0])) <= 1e-9 * abs(whole[0])


def test_grouped_checksum_of_checksums():
    """the grand total equals the sum over group sums (full-size-friendly
    size-independent property)."""
    n = 150_000
    rng = np.random.default_rng(69)
    keys = [b"G%02d" % v for v in rng.integers(0, 40, n)]
    vals = rng.random(n)
    t = po.OracleTable([po.T_STRING, po.T_DOUBLE])
    for s in range(0, n, 50_000):
        e = min(n, s + 50_000)
        t.add_batch(e - s, [po.encode(po.T_STRING, po.ENC_DICT, keys[s:e]),
                            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, vals[s:e])])
    grows = po.result_rows(t.query(po.make_plan(
        group_cols=[0], aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])))
    krows = po.result_rows(t.query(po.make_plan(
        aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])))
    gsum = sum(v[0] for _, v in grows)
    gcnt = sum(v[1] for _, v in grows)
    assert gcnt == krows[0][1][1] == float(n)
    assert abs(gsum - krows[0][1][0]) <= 1e-9 * abs(gsum)
