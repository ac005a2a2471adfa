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
