"""Generate the committed golden fixtures under tests/golden/.

Run in the build container (where /root/reference exists):
    python tests/golden/make_fixtures.py

Outputs (committed):
  - vectors.npz        : encoder byte-layout pins — inputs + expected blobs for
                         every (dtype, encoding, nullability) case.  Layout has
                         no golden bytes in the reference (round-trip tests
                         only, ColumnEncodersTest.scala:26-33), so these
                         vectors ARE the byte-level pin for this build.
  - lineitem_batches.bin: the reference's bundled TPCH lineitem.tbl
                         (tests/common/src/main/resources/TPCH/, 30,201 rows)
                         encoded into reference-format column batches, so GPU
                         parity tests can reproduce Snappy_1.out/Snappy_6.out
                         on boxes without /root/reference.
"""
import json
import os
import struct
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO)

from oracle import pyoracle as po          # noqa: E402
from tests import tpch_util as tu          # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))


def vector_cases():
    rng = np.random.default_rng(12345)
    cases = []

    def add(name, dtype, enc, values, valid=None):
        cases.append((name, dtype, enc, values, valid))

    for dtype, name in [(po.T_INT32, "i32"), (po.T_INT64, "i64"),
                        (po.T_DOUBLE, "f64"), (po.T_FLOAT, "f32"),
                        (po.T_INT16, "i16"), (po.T_INT8, "i8")]:
        np_t = po._NP_OF_T[dtype]
        if dtype in (po.T_DOUBLE, po.T_FLOAT):
            v = (rng.random(257) * 100 - 50).astype(np_t)
        else:
            lim = min(10**6, np.iinfo(np_t).max)
            v = rng.integers(0, lim, 257).astype(np_t)
        add(f"unc_{name}", dtype, po.ENC_UNCOMPRESSED, v)
        valid = (rng.random(257) >= 0.25).astype(np.uint8)
        add(f"unc_{name}_null", dtype, po.ENC_UNCOMPRESSED, v, valid)
    # RLE
    for dtype, name in [(po.T_INT16, "i16"), (po.T_INT32, "i32"), (po.T_INT64, "i64")]:
        np_t = po._NP_OF_T[dtype]
        v = np.repeat(rng.integers(0, 7, 40), rng.integers(1, 9, 40))[:200].astype(np_t)
        add(f"rle_{name}", dtype, po.ENC_RLE, v)
        valid = (rng.random(len(v)) >= 0.2).astype(np.uint8)
        add(f"rle_{name}_null", dtype, po.ENC_RLE, v, valid)
    # dictionary int
    for enc, ename in [(po.ENC_DICT, "dict"), (po.ENC_BIGDICT, "bigdict")]:
        v = rng.integers(0, 11, 300).astype(np.int32)
        add(f"{ename}_i32", po.T_INT32, enc, v)
        valid = (rng.random(300) >= 0.3).astype(np.uint8)
        add(f"{ename}_i32_null", po.T_INT32, enc, v, valid)
    # bool bitset
    v = rng.integers(0, 2, 300).astype(np.uint8)
    add("boolbs", po.T_BOOL, po.ENC_BOOLBITSET, v)
    valid = (rng.random(300) >= 0.3).astype(np.uint8)
    add("boolbs_null", po.T_BOOL, po.ENC_BOOLBITSET, v, valid)
    # strings
    pool = [b"A", b"N", b"R", b"", b"medium-string", b"Z" * 47]
    sv = [pool[rng.integers(0, len(pool))] for _ in range(300)]
    svn = [None if rng.random() < 0.25 else s for s in sv]
    for enc, ename in [(po.ENC_UNCOMPRESSED, "unc"), (po.ENC_DICT, "dict"),
                       (po.ENC_BIGDICT, "bigdict")]:
        add(f"{ename}_str", po.T_STRING, enc, sv)
        add(f"{ename}_str_null", po.T_STRING, enc, svn)
    add("rle_str", po.T_STRING, po.ENC_RLE, sorted(sv))
    return cases


def pack_strings(values):
    lens = np.array([-1 if v is None else len(v) for v in values], dtype=np.int32)
    payload = b"".join(v for v in values if v is not None)
    return lens, np.frombuffer(payload, dtype=np.uint8) if payload else np.zeros(0, np.uint8)


def make_vectors():
    out = {}
    manifest = []
    for name, dtype, enc, values, valid in vector_cases():
        if dtype == po.T_STRING:
            vv = [(b"" if v is None else v) for v in values]
            va = np.array([0 if v is None else 1 for v in values], dtype=np.uint8) \
                if any(v is None for v in values) else valid
            blob = po.encode(dtype, enc, values, valid)
            lens, payload = pack_strings(values)
            out[f"{name}__len"] = lens
            out[f"{name}__payload"] = payload
        else:
            blob = po.encode(dtype, enc, values, valid)
            out[f"{name}__values"] = values
        if valid is not None:
            out[f"{name}__valid"] = valid
        out[f"{name}__blob"] = np.frombuffer(blob, dtype=np.uint8)
        manifest.append(dict(name=name, dtype=int(dtype), enc=int(enc),
                             count=len(values)))
    out["__manifest"] = np.frombuffer(json.dumps(manifest).encode(), dtype=np.uint8)
    np.savez_compressed(os.path.join(HERE, "vectors.npz"), **out)
    print(f"vectors.npz: {len(manifest)} cases")


def write_batches(path, batches):
    with open(path, "wb") as f:
        f.write(struct.pack("<i", len(batches)))
        for num_rows, cols, stats in batches:
            f.write(struct.pack("<ii", num_rows, len(cols)))
            for blob in cols:
                f.write(struct.pack("<q", len(blob)))
                f.write(blob)
            s = stats or b""
            f.write(struct.pack("<q", len(s)))
            f.write(s)


def read_batches(path):
    out = []
    with open(path, "rb") as f:
        (nb,) = struct.unpack("<i", f.read(4))
        for _ in range(nb):
            num_rows, nc = struct.unpack("<ii", f.read(8))
            cols = []
            for _ in range(nc):
                (ln,) = struct.unpack("<q", f.read(8))
                cols.append(f.read(ln))
            (ln,) = struct.unpack("<q", f.read(8))
            stats = f.read(ln) if ln else None
            out.append((num_rows, cols, stats))
    return out


def make_lineitem():
    data = tu.load_lineitem_tbl()
    batches = tu.encode_lineitem_batches(data, 4096)
    write_batches(os.path.join(HERE, "lineitem_batches.bin"), batches)
    n = sum(b[0] for b in batches)
    print(f"lineitem_batches.bin: {len(batches)} batches, {n} rows")


if __name__ == "__main__":
    make_vectors()
    make_lineitem()
