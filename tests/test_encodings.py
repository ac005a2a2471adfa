"""Round-trip + edge-case tests of the column byte formats (oracle encoders/
decoders), mirroring the reference's encoder property tests
(ColumnEncodersTest.scala:26-33, ColumnTablesTestBase.runAllTypesTest) which
pin behavior by round-trip.  Byte-level layout itself is pinned by the golden
fixtures committed under tests/golden/ (see test_golden_fixtures.py)."""
import numpy as np
import pytest

from oracle import pyoracle as po

RNG = np.random.default_rng(42)


def _rand_valid(n, frac_null):
    if frac_null == 0:
        return None
    v = (RNG.random(n) >= frac_null).astype(np.uint8)
    return v


FIXED_CASES = [
    (po.T_INT32, po.ENC_UNCOMPRESSED), (po.T_INT64, po.ENC_UNCOMPRESSED),
    (po.T_DOUBLE, po.ENC_UNCOMPRESSED), (po.T_FLOAT, po.ENC_UNCOMPRESSED),
    (po.T_INT16, po.ENC_UNCOMPRESSED), (po.T_INT8, po.ENC_UNCOMPRESSED),
    (po.T_BOOL, po.ENC_UNCOMPRESSED),
    (po.T_INT16, po.ENC_RLE), (po.T_INT32, po.ENC_RLE), (po.T_INT64, po.ENC_RLE),
    (po.T_INT32, po.ENC_DICT), (po.T_INT64, po.ENC_DICT),
    (po.T_INT32, po.ENC_BIGDICT), (po.T_INT64, po.ENC_BIGDICT),
    (po.T_BOOL, po.ENC_BOOLBITSET),
]


def _gen_fixed(dtype, n, few_distinct):
    if dtype == po.T_DOUBLE:
        return RNG.random(n)
    if dtype == po.T_FLOAT:
        return RNG.random(n).astype(np.float32)
    if dtype == po.T_BOOL:
        return RNG.integers(0, 2, n).astype(np.uint8)
    hi = 5 if few_distinct else 10**6
    info = {po.T_INT32: np.int32, po.T_INT64: np.int64,
            po.T_INT16: np.int16, po.T_INT8: np.uint8}[dtype]
    lim = min(hi, np.iinfo(info).max)
    return RNG.integers(0, lim, n).astype(info)


@pytest.mark.parametrize("dtype,enc", FIXED_CASES)
@pytest.mark.parametrize("frac_null", [0.0, 0.3])
def test_roundtrip_fixed(dtype, enc, frac_null):
    n = 1000
    few = enc in (po.ENC_RLE, po.ENC_DICT, po.ENC_BIGDICT)
    vals = _gen_fixed(dtype, n, few)
    valid = _rand_valid(n, frac_null)
    blob = po.encode(dtype, enc, vals, valid)
    got, gvalid = po.decode(dtype, blob, n)
    exp_valid = np.ones(n, dtype=np.uint8) if valid is None else valid
    assert np.array_equal(gvalid, exp_valid)
    m = exp_valid.astype(bool)
    if dtype in (po.T_DOUBLE, po.T_FLOAT):
        assert np.array_equal(got[m], vals[m])  # bit-exact round trip
    else:
        assert np.array_equal(got[m].astype(np.int64), vals[m].astype(np.int64))


@pytest.mark.parametrize("enc", [po.ENC_UNCOMPRESSED, po.ENC_DICT, po.ENC_BIGDICT, po.ENC_RLE])
@pytest.mark.parametrize("frac_null", [0.0, 0.3])
def test_roundtrip_strings(enc, frac_null):
    n = 500
    pool = [b"A", b"N", b"R", b"", b"longer-string-value", b"x" * 40]
    vals = [pool[RNG.integers(0, len(pool))] for _ in range(n)]
    if enc == po.ENC_RLE:  # RLE wants runs
        vals = sorted(vals)
    if frac_null > 0:
        vals = [None if RNG.random() < frac_null else v for v in vals]
    blob = po.encode(po.T_STRING, enc, vals)
    got, _ = po.decode(po.T_STRING, blob, n)
    assert got == vals


def test_empty_column():
    blob = po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, np.zeros(0, dtype=np.int32))
    got, valid = po.decode(po.T_INT32, blob, 0)
    assert len(got) == 0
    blob = po.encode(po.T_STRING, po.ENC_DICT, [])
    got, _ = po.decode(po.T_STRING, blob, 0)
    assert got == []


def test_all_null_column():
    n = 128
    valid = np.zeros(n, dtype=np.uint8)
    blob = po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, np.zeros(n), valid)
    got, gvalid = po.decode(po.T_DOUBLE, blob, n)
    assert not gvalid.any()
    # header: typeId 0, nullBytes = 2 words * 8, no body
    assert len(blob) == 8 + 16


def test_header_layout_bytes():
    """Pin the blob header byte-for-byte (ColumnEncoding.scala:37-53,764-773,
    1080-1092: [int32 typeId][int32 nullBytes multiple of 8][words][body] LE)."""
    vals = np.array([7, -1, 123456], dtype=np.int32)
    blob = po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, vals)
    assert blob[:8] == (0).to_bytes(4, "little") + (0).to_bytes(4, "little")
    assert blob[8:] == vals.tobytes()
    valid = np.array([1, 0, 1], dtype=np.uint8)
    blob = po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, vals, valid)
    # null bitset: bit 1 set => word 0x2; only non-null values in body
    assert blob[:8] == (0).to_bytes(4, "little") + (8).to_bytes(4, "little")
    assert blob[8:16] == (2).to_bytes(8, "little")
    assert blob[16:] == vals[[0, 2]].tobytes()


def test_dictionary_layout_bytes():
    """Pin dictionary layout (DictionaryEncoding.scala:85-160): [int32 n]
    [entries: int32 len + utf8][int16 indexes], null index == numElements."""
    vals = [b"N", b"A", b"N", None, b"A"]
    blob = po.encode(po.T_STRING, po.ENC_DICT, vals)
    # header: typeId 2, nullBytes 8, one null word (bit 3)
    assert blob[0:4] == (2).to_bytes(4, "little")
    assert blob[4:8] == (8).to_bytes(4, "little")
    assert int.from_bytes(blob[8:16], "little") == 1 << 3
    body = blob[16:]
    assert body[0:4] == (2).to_bytes(4, "little")          # numElements
    assert body[4:8] == (1).to_bytes(4, "little") and body[8:9] == b"N"
    assert body[9:13] == (1).to_bytes(4, "little") and body[13:14] == b"A"
    idx = np.frombuffer(body[14:], dtype=np.int16)
    # index array holds only non-null entries (writeIsNull writes no body)
    assert list(idx) == [0, 1, 0, 1]


def test_delete_mask_layout_and_semantics():
    blob = po.encode_delete([0, 2, 5], 6)
    assert blob == b"".join(x.to_bytes(4, "little", signed=True) for x in [0, 6, 3, 0, 2, 5])
    t = po.OracleTable([po.T_INT32])
    col = po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, np.arange(6, dtype=np.int32))
    t.add_batch(6, [col], delete_mask=blob)
    plan = po.make_plan(aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])])
    rows = po.result_rows(t.query(plan))
    # rows 1,3,4 survive: sum=8, count=3
    assert rows[0][1] == [8.0, 3.0]


def test_delete_all_rows():
    t = po.OracleTable([po.T_INT32])
    col = po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, np.arange(4, dtype=np.int32))
    t.add_batch(4, [col], delete_mask=po.encode_delete([0, 1, 2, 3], 4))
    rows = po.result_rows(t.query(po.make_plan(aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])])))
    assert rows[0][1] == [None, 0.0]  # empty SUM is NULL, COUNT 0


def test_update_delta_merge():
    """delta1 overrides delta2 overrides base (UpdatedColumnDecoder.scala:69-115)."""
    n = 10
    base = np.arange(n, dtype=np.int32) * 10
    col = po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, base)
    # delta2 (higher depth, merged earlier) updates pos 2->200, 5->500
    d2 = po.encode_delta(po.T_INT32, po.ENC_UNCOMPRESSED, [2, 5], n,
                         np.array([200, 500], dtype=np.int32))
    # delta1 updates pos 5->555 (overrides delta2), 7->777
    d1 = po.encode_delta(po.T_INT32, po.ENC_UNCOMPRESSED, [5, 7], n,
                         np.array([555, 777], dtype=np.int32))
    t = po.OracleTable([po.T_INT32])
    t.add_batch(-n, [col], deltas=[(d1, d2)])  # negative rows => has deltas
    rows = po.result_rows(t.query(po.make_plan(aggs=[("sum", [(0, 0.0, 1.0)])])))
    expect = float(sum([0, 10, 200, 30, 40, 555, 60, 777, 80, 90]))
    assert rows[0][1][0] == expect


def test_update_delta_null_values():
    """a delta can write NULL over a non-null base value (delta null bitset
    indexes delta entries, ColumnDeltaDecoder.scala:45-58)."""
    n = 4
    base = np.array([1, 2, 3, 4], dtype=np.int32)
    col = po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, base)
    d1 = po.encode_delta(po.T_INT32, po.ENC_UNCOMPRESSED, [1, 3], n,
                         np.array([99, 0], dtype=np.int32),
                         valid=np.array([1, 0], dtype=np.uint8))
    t = po.OracleTable([po.T_INT32])
    t.add_batch(-n, [col], deltas=[(d1, None)])
    rows = po.result_rows(t.query(po.make_plan(
        aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])])))
    # values: 1, 99, 3, NULL -> sum 103, count(*) = 4
    assert rows[0][1] == [103.0, 4.0]


def test_deleted_rows_with_deltas_and_nulls():
    """delete + delta + null interplay on one batch."""
    n = 6
    base = np.array([1, 2, 3, 4, 5, 6], dtype=np.int32)
    valid = np.array([1, 1, 0, 1, 1, 1], dtype=np.uint8)
    col = po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, base, valid)
    d1 = po.encode_delta(po.T_INT32, po.ENC_UNCOMPRESSED, [2, 4], n,
                         np.array([33, 55], dtype=np.int32))
    dm = po.encode_delete([1, 4], n)
    t = po.OracleTable([po.T_INT32])
    t.add_batch(-n, [col], delete_mask=dm, deltas=[(d1, None)])
    rows = po.result_rows(t.query(po.make_plan(
        aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])])))
    # surviving rows 0,2,3,5: values 1, 33(delta over null), 4, 6 -> 44, count 4
    assert rows[0][1] == [44.0, 4.0]


def test_group_by_null_keys():
    keys = [b"a", None, b"b", None, b"a"]
    vals = np.array([1.0, 2.0, 3.0, 4.0, 5.0])
    t = po.OracleTable([po.T_STRING, po.T_DOUBLE])
    t.add_batch(5, [po.encode(po.T_STRING, po.ENC_DICT, keys),
                    po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, vals)])
    rows = po.result_rows(t.query(po.make_plan(
        group_cols=[0], aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])))
    d = {k[0]: v for k, v in rows}
    assert d[b"a".decode()] == [6.0, 2.0]
    assert d[b"b".decode()] == [3.0, 1.0]
    assert d[None] == [6.0, 2.0]


def test_sum_skips_nulls_count_star_does_not():
    n = 8
    v = np.arange(n, dtype=np.float64)
    valid = np.array([1, 0, 1, 0, 1, 1, 1, 0], dtype=np.uint8)
    t = po.OracleTable([po.T_DOUBLE])
    t.add_batch(n, [po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, v, valid)])
    rows = po.result_rows(t.query(po.make_plan(
        aggs=[("sum", [(0, 0.0, 1.0)]), ("avg", [(0, 0.0, 1.0)]), ("count", [])])))
    s = float(v[valid.astype(bool)].sum())
    assert rows[0][1][0] == s
    assert rows[0][1][1] == s / 5
    assert rows[0][1][2] == 8.0


def test_predicate_on_null_is_false():
    n = 4
    v = np.array([1, 2, 3, 4], dtype=np.int32)
    valid = np.array([1, 0, 1, 0], dtype=np.uint8)
    t = po.OracleTable([po.T_INT32])
    t.add_batch(n, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, v, valid)])
    rows = po.result_rows(t.query(po.make_plan(
        preds=[dict(col=0, lo=0)], aggs=[("count", [])])))
    assert rows[0][1][0] == 2.0


def test_config1_vs_numpy():
    """BASELINE config 1 shape: SELECT SUM(d) WHERE i > k, 100K rows."""
    n = 100_000
    i = RNG.integers(0, 10**6, n).astype(np.int32)
    d = RNG.random(n)
    k = int(np.median(i))
    t = po.OracleTable([po.T_INT32, po.T_DOUBLE])
    for s in range(0, n, 8192):
        e = min(n, s + 8192)
        t.add_batch(e - s, [po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, i[s:e]),
                            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, d[s:e])])
    rows = po.result_rows(t.query(po.make_plan(
        preds=[dict(col=0, lo=k, lo_strict=True)],
        aggs=[("sum", [(1, 0.0, 1.0)]), ("count", [])])))
    m = i > k
    assert rows[0][1][1] == float(m.sum())
    assert abs(rows[0][1][0] - d[m].sum()) < 1e-6 * abs(d[m].sum())


def test_rle_layout_bytes():
    """Pin RLE layout: [value][int32 run] pairs (int32: 8B/run),
    RunLengthEncoding.scala:131-143 decoder semantics."""
    v = np.array([5, 5, 5, 9, 9, 5], dtype=np.int32)
    blob = po.encode(po.T_INT32, po.ENC_RLE, v)
    assert blob[0:4] == (1).to_bytes(4, "little")
    assert blob[4:8] == (0).to_bytes(4, "little")
    runs = blob[8:]
    exp = b"".join(x.to_bytes(4, "little") for x in [5, 3, 9, 2, 5, 1])
    assert runs == exp


def test_dictionary_promotion_boundary():
    """encoder promotes to BigDictionary exactly when the dictionary would
    exceed Short.MaxValue entries (DictionaryEncoding.scala:313,338)."""
    import numpy as np
    below = [b"K%06d" % i for i in range(32767)]
    blob = po.encode(po.T_STRING, po.ENC_DICT, below)
    assert int.from_bytes(blob[:4], "little") == po.ENC_DICT
    over = [b"K%06d" % i for i in range(32768)]
    blob2 = po.encode(po.T_STRING, po.ENC_BIGDICT, over)
    assert int.from_bytes(blob2[:4], "little") == po.ENC_BIGDICT
    out, valid = po.decode(po.T_STRING, blob2, len(over))
    assert list(out[:3]) == over[:3] and list(out[-2:]) == over[-2:]


def test_delta_roundtrip_int64_and_int16():
    """update-delta encode/decode for integer widths (value-as-double path)."""
    import numpy as np
    n = 10_000
    for dt, npdt, vals in ((po.T_INT64, np.int64, [2**40, -7]),
                           (po.T_INT16, np.int16, [123, -456])):
        base = np.zeros(n, dtype=npdt)
        pos = np.array([3, 8], dtype=np.int32)
        d = po.encode_delta(dt, po.ENC_UNCOMPRESSED, pos, n,
                            np.array(vals, dtype=npdt))
        t = po.OracleTable([dt])
        t.add_batch(-n, [po.encode(dt, po.ENC_UNCOMPRESSED, base)],
                    deltas=[(d, None)])
        rows = po.result_rows(t.query(po.make_plan(
            aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])])))
        assert rows[0][1][0] == float(sum(vals))
        assert rows[0][1][1] == float(n)


def test_malformed_blobs_rejected():
    """truncated/corrupt blobs must fail loudly, not crash (host put path
    and oracle decoder)."""
    import numpy as np
    from snappydata_amd import abi, engine as se
    good = po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, np.arange(100.0))
    bad_cases = [
        good[:6],                                     # shorter than header
        b"\xff\xff\xff\x7f" + good[4:],               # absurd typeId
        good[:4] + (7).to_bytes(4, "little") + good[8:],   # nullBytes % 8 != 0
        good[:4] + (1 << 20).to_bytes(4, "little"),   # null bytes beyond blob
    ]
    e = se.Engine(device=-1)
    t = e.table_define("tbad", [(abi.T_DOUBLE, False)])
    for i, blob in enumerate(bad_cases):
        try:
            e.batch_put(t, i, 0, 100, [blob])
            raise AssertionError(f"case {i} must be rejected")
        except se.EngineError:
            pass
    # a put that failed must not have appended a batch
    assert e.num_rows(t) == 0
    e.close()

    # the oracle validates lazily (decode at query time): either the put or
    # the first query must fail — never crash or return wrong numbers
    plan = po.make_plan(aggs=[("sum", [(0, 0.0, 1.0)]), ("count", [])])
    for i, blob in enumerate(bad_cases):
        ot = po.OracleTable([po.T_DOUBLE])
        try:
            ot.add_batch(100, [blob])
        except Exception:
            continue
        try:
            ot.query(plan)
            raise AssertionError(f"oracle case {i} must fail at query")
        except AssertionError as ex:
            if "must fail" in str(ex):
                raise
        except Exception:
            pass


def test_truncated_delete_mask_and_delta_rejected():
    import numpy as np
    from snappydata_amd import abi, engine as se
    n = 1000
    blob = po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, np.arange(float(n)))
    dmask = po.encode_delete(np.array([1, 2], dtype=np.int32), n)
    e = se.Engine(device=-1)
    t = e.table_define("tbad2", [(abi.T_DOUBLE, False)])
    try:
        e.batch_put(t, 0, 0, n, [blob], delete_mask=dmask[:10])
        raise AssertionError("truncated delete mask must be rejected")
    except se.EngineError:
        pass
    d1 = po.encode_delta(po.T_DOUBLE, po.ENC_UNCOMPRESSED,
                         np.array([5], dtype=np.int32), n, np.array([1.0]))
    try:
        e.batch_put(t, 1, 0, -n, [blob], deltas=[(d1[:8], None)])
        raise AssertionError("truncated delta must be rejected")
    except se.EngineError:
        pass
    e.close()


def test_corrupt_dictionary_index_rejected():
    """an index past the dictionary (beyond the NULL sentinel) must be
    rejected at put — the scan kernel would otherwise walk off the map."""
    import numpy as np
    from snappydata_amd import abi, engine as se
    keys = [b"A", b"B", b"A", b"C"] * 100
    blob = bytearray(po.encode(po.T_STRING, po.ENC_DICT, keys))
    # index array is the tail: poke one int16 index to 999 (> dict size 3)
    blob[-2:] = (999).to_bytes(2, "little")
    e = se.Engine(device=-1)
    t = e.table_define("tdx", [(abi.T_STRING, False)])
    try:
        e.batch_put(t, 0, 0, len(keys), [bytes(blob)])
        raise AssertionError("corrupt dictionary index must be rejected")
    except se.EngineError:
        pass
    # the untouched blob still loads
    e.batch_put(t, 1, 0, len(keys), [po.encode(po.T_STRING, po.ENC_DICT, keys)])
    assert e.num_rows(t) == len(keys)
    e.close()
