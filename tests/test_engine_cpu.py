"""CPU-side tests of the PRODUCT engine host code (no GPU needed):
builder encoding (cross-checked against the oracle byte-for-byte), datagen
determinism, and loud failure of the query path without a GPU.

The engine in host-only mode (device=-1 when no HIP device) stores blobs in
host shadow memory so tests can fetch them back; queries are REFUSED with
SN_ERR_NOGPU — the engine never silently computes on CPU.
"""
import numpy as np
import pytest

from oracle import pyoracle as po
from snappydata_amd import abi, engine as se


@pytest.fixture(scope="module")
def eng():
    e = se.Engine(device=-1)
    yield e
    e.close()


def test_engine_arch():
    assert se.lib().sn_engine_arch() == b"gfx950"


def test_builder_matches_oracle_encoder_numeric(eng):
    rng = np.random.default_rng(3)
    n = 10000
    i32 = rng.integers(0, 10**6, n).astype(np.int32)
    f64 = rng.random(n)
    t = eng.table_define("t_num", [(abi.T_INT32, False), (abi.T_DOUBLE, False)])
    got = eng.ingest_columns(t, [{"data": i32}, {"data": f64}], n, batch_rows=4096)
    assert got == n
    assert eng.num_batches(t) == (n + 4095) // 4096
    # engine-encoded blobs must equal the oracle's byte-for-byte (same spec)
    for b in range(eng.num_batches(t)):
        s, e_ = b * 4096, min(n, (b + 1) * 4096)
        assert eng.get_blob(t, b, 0) == po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, i32[s:e_])
        assert eng.get_blob(t, b, 1) == po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, f64[s:e_])


def test_builder_matches_oracle_encoder_strings_and_nulls(eng):
    rng = np.random.default_rng(4)
    n = 3000
    pool = [b"A", b"N", b"R"]
    vals = [pool[rng.integers(0, 3)] for _ in range(n)]
    valid = (rng.random(n) >= 0.2).astype(np.uint8)
    f64 = rng.random(n)
    t = eng.table_define("t_str", [(abi.T_STRING, True), (abi.T_DOUBLE, True)])
    payload = b"".join(v if valid[i] else b"" for i, v in enumerate(vals))
    lens = np.array([len(v) if valid[i] else 0 for i, v in enumerate(vals)],
                    dtype=np.int32)
    eng.ingest_columns(t, [{"data": payload, "lens": lens, "valid": valid},
                           {"data": f64, "valid": valid}], n, batch_rows=1024)
    ovals = [vals[i] if valid[i] else None for i in range(n)]
    for b in range(eng.num_batches(t)):
        s, e_ = b * 1024, min(n, (b + 1) * 1024)
        assert eng.get_blob(t, b, 0) == po.encode(po.T_STRING, po.ENC_DICT, ovals[s:e_])
        assert eng.get_blob(t, b, 1) == po.encode(
            po.T_DOUBLE, po.ENC_UNCOMPRESSED, f64[s:e_], valid[s:e_])


def test_engine_blobs_decode_by_oracle(eng):
    """Product-encoded lineitem batches decode correctly through the oracle."""
    n = 5000
    d = se.gen_lineitem_arrays(0, n, seed=42)
    t = eng.table_define("t_li", [(abi.T_DOUBLE, False)] * 4 +
                         [(abi.T_STRING, False)] * 2 + [(abi.T_INT32, False)])
    eng.ingest_columns(t, [
        {"data": d["qty"]}, {"data": d["ep"]}, {"data": d["disc"]}, {"data": d["tax"]},
        {"data": b"".join(d["rf"]), "lens": np.ones(n, dtype=np.int32)},
        {"data": b"".join(d["ls"]), "lens": np.ones(n, dtype=np.int32)},
        {"data": d["ship"]}], n, batch_rows=2048)
    for b in range(eng.num_batches(t)):
        s, e_ = b * 2048, min(n, (b + 1) * 2048)
        vals, _ = po.decode(po.T_DOUBLE, eng.get_blob(t, b, 1), e_ - s)
        assert np.array_equal(vals, d["ep"][s:e_])
        svals, _ = po.decode(po.T_STRING, eng.get_blob(t, b, 4), e_ - s)
        assert svals == d["rf"][s:e_]
        ivals, _ = po.decode(po.T_INT32, eng.get_blob(t, b, 6), e_ - s)
        assert np.array_equal(ivals, d["ship"][s:e_])


def test_datagen_deterministic():
    a = se.gen_lineitem_arrays(1000, 100, seed=7)
    b = se.gen_lineitem_arrays(1000, 100, seed=7)
    assert np.array_equal(a["qty"], b["qty"])
    assert np.array_equal(a["ship"], b["ship"])
    c = se.gen_lineitem_arrays(1000, 100, seed=8)
    assert not np.array_equal(a["ship"], c["ship"])
    # distribution sanity for Q6 selectivity (~1.9%)
    d = se.gen_lineitem_arrays(0, 200000, seed=42)
    from tests import tpch_util as tu
    m = ((d["ship"] >= tu.days(1994, 1, 1)) & (d["ship"] < tu.days(1995, 1, 1)) &
         (d["disc"] >= 0.05) & (d["disc"] <= 0.07) & (d["qty"] < 24))
    sel = m.mean()
    assert 0.01 < sel < 0.03, sel


def test_query_without_gpu_fails_loudly(eng):
    t = eng.table_define("t_q", [(abi.T_DOUBLE, False)])
    eng.ingest_columns(t, [{"data": np.arange(10.0)}], 10)
    with pytest.raises(se.EngineError):
        eng.query(abi.make_plan(table=t, aggs=[("sum", [(0, 0.0, 1.0)])]))
    assert "no HIP device" in se.last_error()


def test_shard_filtering_cpu():
    """bucket % shard_count routing (one process per GPU): each shard keeps a
    disjoint batch subset; union covers everything."""
    n = 10000
    d = se.gen_lineitem_arrays(0, n, seed=1)
    counts = []
    for rank in range(2):
        e = se.Engine(device=-1, shard_rank=rank, shard_count=2)
        t = e.table_define("li", [(abi.T_DOUBLE, False)] * 4 +
                           [(abi.T_STRING, False)] * 2 + [(abi.T_INT32, False)])
        got = e.datagen_lineitem(t, n, seed=1, batch_rows=1024)
        assert got >= 0
        counts.append(e.num_rows(t))
        e.close()
    assert sum(counts) == n
    assert all(c > 0 for c in counts)


def test_batch_mutate_hostmode_no_crash():
    """sn_batch_mutate on a host-only engine (no GPU): state bookkeeping
    works, unknown batches fail loudly, get_blob keeps base bytes."""
    import numpy as np
    from oracle import pyoracle as po
    n = 5_000
    f64 = np.arange(n, dtype=np.float64)
    e = se.Engine(device=-1)
    t = e.table_define("tm", [(abi.T_DOUBLE, False)])
    blob = po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, f64)
    e.batch_put(t, 5, 0, n, [blob])
    d1 = se.encode_update_delta(abi.T_DOUBLE, np.array([1], dtype=np.int32),
                                n, np.array([99.0]))
    e.batch_mutate(t, 5, 0, deltas=[(d1, None)])
    assert e.get_blob(t, 0, 0) == blob          # base bytes untouched
    try:
        e.batch_mutate(t, 6, 0, deltas=[(d1, None)])
        raise AssertionError("unknown uuid must fail")
    except se.EngineError as ex:
        assert ex.code == abi.SN_ERR_BADARG if hasattr(abi, "SN_ERR_BADARG") else True
    e.close()
