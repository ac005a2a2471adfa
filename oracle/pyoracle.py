"""ctypes wrapper over the CPU oracle (oracle/libsn_oracle.so).

TEST INFRASTRUCTURE ONLY — importable solely from tests/, __graft_entry__.smoke()
and bench.py's cpu_baseline leg.  The product path (snappydata_amd/) must never
import this module.

The structs here mirror include/snappy_engine.h exactly (ctypes restatement).
"""
import ctypes as C
import os
import subprocess

_DIR = os.path.dirname(os.path.abspath(__file__))

SN_MAX_PREDS = 8
SN_MAX_AGGS = 12
SN_MAX_GROUPS = 2
SN_MAX_FACTORS = 3
SN_MAX_GROUP_SLOTS = 1024
SN_KEY_MAX = 48

# sn_type_t
T_INT32, T_INT64, T_DOUBLE, T_STRING, T_BOOL, T_INT16, T_INT8, T_FLOAT = range(8)
# encodings
ENC_UNCOMPRESSED, ENC_RLE, ENC_DICT, ENC_BIGDICT, ENC_BOOLBITSET = range(5)
AGG_SUM, AGG_COUNT_STAR, AGG_AVG, AGG_MIN, AGG_MAX = range(5)


class SnBuf(C.Structure):
    _fields_ = [("data", C.c_void_p), ("len", C.c_int64)]


class SnPred(C.Structure):
    _fields_ = [("col", C.c_int32), ("_pad", C.c_int32),
                ("lo_d", C.c_double), ("hi_d", C.c_double),
                ("lo_i", C.c_int64), ("hi_i", C.c_int64),
                ("has_lo", C.c_uint8), ("has_hi", C.c_uint8),
                ("lo_strict", C.c_uint8), ("hi_strict", C.c_uint8),
                ("_pad2", C.c_uint8 * 4),
                ("str_eq", C.c_char_p), ("str_len", C.c_int32),
                ("_pad3", C.c_int32),
                ("in_i", C.POINTER(C.c_int64)),
                ("in_s", C.POINTER(C.c_char_p)),
                ("in_s_len", C.POINTER(C.c_int32)),
                ("in_n", C.c_int32), ("_pad6", C.c_int32)]


class SnFactor(C.Structure):
    _fields_ = [("col", C.c_int32), ("_pad", C.c_int32),
                ("add", C.c_double), ("mul", C.c_double)]


class SnAgg(C.Structure):
    _fields_ = [("kind", C.c_int32), ("nfactors", C.c_int32),
                ("factors", SnFactor * SN_MAX_FACTORS)]


class SnPlan(C.Structure):
    _fields_ = [("table", C.c_int32), ("npreds", C.c_int32),
                ("preds", SnPred * SN_MAX_PREDS),
                ("ngroup", C.c_int32),
                ("group_cols", C.c_int32 * SN_MAX_GROUPS),
                ("naggs", C.c_int32), ("_pad", C.c_int32),
                ("aggs", SnAgg * SN_MAX_AGGS),
                ("join_dim", C.c_int32), ("join_fact_col", C.c_int32),
                ("join_mode", C.c_int32), ("_pad4", C.c_int32)]


class SnResult(C.Structure):
    _fields_ = [("nrows", C.c_int32), ("ngroup", C.c_int32),
                ("naggs", C.c_int32), ("_pad", C.c_int32),
                ("keys", (C.c_char * SN_KEY_MAX) * SN_MAX_GROUPS * SN_MAX_GROUP_SLOTS),
                ("key_is_null", (C.c_uint8 * SN_MAX_GROUPS) * SN_MAX_GROUP_SLOTS),
                ("vals", (C.c_double * SN_MAX_AGGS) * SN_MAX_GROUP_SLOTS),
                ("val_is_null", (C.c_uint8 * SN_MAX_AGGS) * SN_MAX_GROUP_SLOTS),
                ("rows_scanned", C.c_int64), ("rows_passed", C.c_int64),
                ("batches_seen", C.c_int64), ("batches_skipped", C.c_int64)]


def _build():
    so = os.path.join(_DIR, "libsn_oracle.so")
    src = os.path.join(_DIR, "sn_oracle.c")
    if (not os.path.exists(so)) or os.path.getmtime(so) < os.path.getmtime(src):
        subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)
    return so


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = C.CDLL(_build())
        _lib.sno_table_create.restype = C.c_void_p
        _lib.sno_table_create.argtypes = [C.c_int32, C.POINTER(C.c_int32), C.POINTER(C.c_uint8)]
        _lib.sno_table_destroy.argtypes = [C.c_void_p]
        _lib.sno_table_add_batch.restype = C.c_int32
        _lib.sno_table_add_batch.argtypes = [C.c_void_p, C.c_int32, C.POINTER(SnBuf),
                                             C.POINTER(SnBuf), C.POINTER(SnBuf), C.POINTER(SnBuf)]
        _lib.sno_table_set_dim.restype = C.c_int32
        _lib.sno_table_set_dim.argtypes = [C.c_void_p, C.POINTER(C.c_int64),
                                           C.c_int64, C.c_char_p,
                                           C.POINTER(C.c_int32)]
        _lib.sno_query.restype = C.c_int32
        _lib.sno_query.argtypes = [C.c_void_p, C.POINTER(SnPlan), C.POINTER(SnResult), C.c_int32]
        _lib.sno_query_groups.restype = C.c_int64
        _lib.sno_query_groups.argtypes = [C.c_void_p, C.POINTER(SnPlan),
                                          C.c_int32, C.c_int64, C.c_char_p,
                                          C.POINTER(C.c_uint8),
                                          C.POINTER(C.c_double),
                                          C.POINTER(C.c_uint8)]
        _lib.sno_encode.restype = C.c_int64
        _lib.sno_encode.argtypes = [C.c_int32, C.c_int32, C.c_int32, C.c_void_p,
                                    C.POINTER(C.c_int32), C.POINTER(C.c_uint8),
                                    C.c_int32, C.c_void_p, C.c_int64]
        _lib.sno_encode_delete.restype = C.c_int64
        _lib.sno_encode_delete.argtypes = [C.POINTER(C.c_int32), C.c_int32, C.c_int32,
                                           C.c_void_p, C.c_int64]
        _lib.sno_encode_delta.restype = C.c_int64
        _lib.sno_encode_delta.argtypes = [C.c_int32, C.c_int32, C.c_int32,
                                          C.POINTER(C.c_int32), C.c_int32, C.c_int32,
                                          C.c_void_p, C.POINTER(C.c_int32), C.POINTER(C.c_uint8),
                                          C.c_void_p, C.c_int64]
        _lib.sno_encode_stats.restype = C.c_int64
        _lib.sno_encode_stats.argtypes = [C.c_int32, C.POINTER(C.c_int32), C.c_int32,
                                          C.POINTER(C.c_double), C.POINTER(C.c_double),
                                          C.POINTER(C.c_int64), C.POINTER(C.c_int64),
                                          C.POINTER(C.c_int32), C.POINTER(C.c_uint8),
                                          C.c_void_p, C.c_int64]
        _lib.sno_decode.restype = C.c_int32
        _lib.sno_decode.argtypes = [C.c_int32, C.c_void_p, C.c_int64, C.c_int32,
                                    C.c_void_p, C.c_int64,
                                    C.POINTER(C.c_int32), C.POINTER(C.c_uint8)]
    return _lib


import numpy as np

_NP_OF_T = {T_INT32: np.int32, T_INT64: np.int64, T_DOUBLE: np.float64,
            T_BOOL: np.uint8, T_INT16: np.int16, T_INT8: np.uint8, T_FLOAT: np.float32}


def encode(dtype, encoding, values, valid=None):
    """Encode one column blob.  values: numpy array (fixed types) or list of
    (bytes|None) for strings.  valid: optional numpy uint8 array (1=valid).
    Returns bytes."""
    L = lib()
    if dtype == T_STRING:
        lens = np.array([0 if v is None else len(v) for v in values], dtype=np.int32)
        blob = b"".join(v for v in values if v is not None or True) if False else \
            b"".join((v or b"") for v in values)
        count = len(values)
        if valid is None and any(v is None for v in values):
            valid = np.array([0 if v is None else 1 for v in values], dtype=np.uint8)
        buf = np.frombuffer(blob, dtype=np.uint8) if blob else np.zeros(0, dtype=np.uint8)
        vp = buf.ctypes.data_as(C.c_void_p)
        lp = lens.ctypes.data_as(C.POINTER(C.c_int32))
    else:
        arr = np.ascontiguousarray(values, dtype=_NP_OF_T[dtype])
        count = len(arr)
        vp = arr.ctypes.data_as(C.c_void_p)
        lp = None
    vdp = valid.ctypes.data_as(C.POINTER(C.c_uint8)) if valid is not None else None
    cap = 64 + count * 32 + (len(blob) * 2 if dtype == T_STRING else 0) + \
        (int(arr.nbytes) * 2 if dtype != T_STRING else 0)
    out = (C.c_uint8 * cap)()
    n = L.sno_encode(dtype, encoding, 1, vp, lp, vdp, count, out, cap)
    assert n >= 0, f"sno_encode failed: {n}"
    return bytes(bytearray(out)[:n])


def encode_delete(positions, num_base_rows):
    L = lib()
    pos = np.ascontiguousarray(positions, dtype=np.int32)
    cap = 16 + 4 * len(pos)
    out = (C.c_uint8 * cap)()
    n = L.sno_encode_delete(pos.ctypes.data_as(C.POINTER(C.c_int32)), len(pos),
                            num_base_rows, out, cap)
    assert n >= 0
    return bytes(bytearray(out)[:n])


def encode_delta(dtype, encoding, positions, num_base_rows, values, valid=None):
    L = lib()
    pos = np.ascontiguousarray(positions, dtype=np.int32)
    if dtype == T_STRING:
        lens = np.array([0 if v is None else len(v) for v in values], dtype=np.int32)
        blob = b"".join((v or b"") for v in values)
        if valid is None and any(v is None for v in values):
            valid = np.array([0 if v is None else 1 for v in values], dtype=np.uint8)
        buf = np.frombuffer(blob, dtype=np.uint8) if blob else np.zeros(0, dtype=np.uint8)
        vp = buf.ctypes.data_as(C.c_void_p)
        lp = lens.ctypes.data_as(C.POINTER(C.c_int32))
        extra = len(blob) * 2
    else:
        arr = np.ascontiguousarray(values, dtype=_NP_OF_T[dtype])
        vp = arr.ctypes.data_as(C.c_void_p)
        lp = None
        extra = int(arr.nbytes) * 2
    vdp = valid.ctypes.data_as(C.POINTER(C.c_uint8)) if valid is not None else None
    cap = 128 + len(pos) * 36 + extra
    out = (C.c_uint8 * cap)()
    n = L.sno_encode_delta(dtype, encoding, 1, pos.ctypes.data_as(C.POINTER(C.c_int32)),
                           len(pos), num_base_rows, vp, lp, vdp, out, cap)
    assert n >= 0, f"sno_encode_delta failed: {n}"
    return bytes(bytearray(out)[:n])


def encode_stats(dtypes, batch_count_signed, lower, upper, null_counts=None,
                 has_bounds=None):
    """lower/upper: per-column python numbers (ints for int cols, floats for
    double cols); entries for string cols ignored unless has_bounds says so."""
    L = lib()
    nc = len(dtypes)
    dts = np.array(dtypes, dtype=np.int32)
    lo_d = np.zeros(nc); hi_d = np.zeros(nc)
    lo_i = np.zeros(nc, dtype=np.int64); hi_i = np.zeros(nc, dtype=np.int64)
    hb = np.ones(nc, dtype=np.uint8) if has_bounds is None else \
        np.ascontiguousarray(has_bounds, dtype=np.uint8)
    for c, dt in enumerate(dtypes):
        if dt == T_STRING:
            hb[c] = 0
            continue
        if not hb[c]:
            continue
        if dt in (T_DOUBLE, T_FLOAT):
            lo_d[c] = float(lower[c]); hi_d[c] = float(upper[c])
        else:
            lo_i[c] = int(lower[c]); hi_i[c] = int(upper[c])
    ncnt = np.zeros(nc, dtype=np.int32) if null_counts is None else \
        np.ascontiguousarray(null_counts, dtype=np.int32)
    cap = 64 + nc * 64
    out = (C.c_uint8 * cap)()
    n = L.sno_encode_stats(nc, dts.ctypes.data_as(C.POINTER(C.c_int32)),
                           batch_count_signed,
                           lo_d.ctypes.data_as(C.POINTER(C.c_double)),
                           hi_d.ctypes.data_as(C.POINTER(C.c_double)),
                           lo_i.ctypes.data_as(C.POINTER(C.c_int64)),
                           hi_i.ctypes.data_as(C.POINTER(C.c_int64)),
                           ncnt.ctypes.data_as(C.POINTER(C.c_int32)),
                           hb.ctypes.data_as(C.POINTER(C.c_uint8)), out, cap)
    assert n >= 0
    return bytes(bytearray(out)[:n])


def decode(dtype, blob, count):
    """Round-trip check helper.  Returns (values, valid) where values is a
    numpy array (or list of bytes for strings)."""
    L = lib()
    b = np.frombuffer(blob, dtype=np.uint8)
    valid = np.zeros(count, dtype=np.uint8)
    if dtype == T_STRING:
        cap = max(1, len(blob) * 2 + count * 256)
        out = (C.c_uint8 * cap)()
        lens = np.zeros(count, dtype=np.int32)
        rc = L.sno_decode(dtype, b.ctypes.data_as(C.c_void_p), len(blob), count,
                          out, cap, lens.ctypes.data_as(C.POINTER(C.c_int32)),
                          valid.ctypes.data_as(C.POINTER(C.c_uint8)))
        assert rc == 0, rc
        vals, off = [], 0
        raw = bytes(bytearray(out))
        for i in range(count):
            if not valid[i]:
                vals.append(None)
            else:
                vals.append(raw[off:off + lens[i]])
                off += lens[i]
        return vals, valid
    arr = np.zeros(count, dtype=_NP_OF_T[dtype])
    rc = L.sno_decode(dtype, b.ctypes.data_as(C.c_void_p), len(blob), count,
                      arr.ctypes.data_as(C.c_void_p), arr.nbytes, None,
                      valid.ctypes.data_as(C.POINTER(C.c_uint8)))
    assert rc == 0, rc
    return arr, valid


class OracleTable:
    def __init__(self, dtypes, nullable=None):
        self.dtypes = list(dtypes)
        nc = len(self.dtypes)
        dts = np.array(self.dtypes, dtype=np.int32)
        nl = np.array(nullable if nullable is not None else [1] * nc, dtype=np.uint8)
        self._h = lib().sno_table_create(nc, dts.ctypes.data_as(C.POINTER(C.c_int32)),
                                         nl.ctypes.data_as(C.POINTER(C.c_uint8)))
        self._keep = []  # keep blob references alive

    def add_batch(self, num_rows, col_blobs, stats=None, delete_mask=None, deltas=None):
        """col_blobs: list of bytes.  deltas: optional list of ncols*(d1,d2)
        tuples of bytes or None.  num_rows may be negative (has deltas)."""
        nc = len(self.dtypes)
        bufs = (SnBuf * nc)()
        arrs = []
        for i, blob in enumerate(col_blobs):
            a = np.frombuffer(blob, dtype=np.uint8)
            arrs.append(a)
            bufs[i].data = a.ctypes.data
            bufs[i].len = len(blob)
        sb = SnBuf()
        if stats:
            sa = np.frombuffer(stats, dtype=np.uint8)
            arrs.append(sa)
            sb.data, sb.len = sa.ctypes.data, len(stats)
        db = SnBuf()
        if delete_mask:
            da = np.frombuffer(delete_mask, dtype=np.uint8)
            arrs.append(da)
            db.data, db.len = da.ctypes.data, len(delete_mask)
        dl = None
        if deltas is not None:
            dl = (SnBuf * (nc * 2))()
            for i, pair in enumerate(deltas):
                for j, d in enumerate(pair):
                    if d:
                        a = np.frombuffer(d, dtype=np.uint8)
                        arrs.append(a)
                        dl[i * 2 + j].data, dl[i * 2 + j].len = a.ctypes.data, len(d)
        self._keep.append(arrs)
        rc = lib().sno_table_add_batch(self._h, num_rows, bufs,
                                       C.byref(sb) if stats else None,
                                       C.byref(db) if delete_mask else None,
                                       dl)
        assert rc == 0, f"add_batch failed: {rc}"

    def set_dim(self, keys, attrs=None):
        """Register the broadcast dimension: keys int64 array; attrs optional
        list of bytes (per key)."""
        keys = np.ascontiguousarray(keys, dtype=np.int64)
        payload, lens = None, None
        if attrs is not None:
            lens = np.array([len(a) for a in attrs], dtype=np.int32)
            payload = b"".join(attrs)
        rc = lib().sno_table_set_dim(
            self._h, keys.ctypes.data_as(C.POINTER(C.c_int64)), len(keys),
            payload, lens.ctypes.data_as(C.POINTER(C.c_int32)) if lens is not None else None)
        assert rc == 0, rc

    def query(self, plan, nthreads=1):
        res = SnResult()
        rc = lib().sno_query(self._h, C.byref(plan), C.byref(res), nthreads)
        assert rc == 0, f"sno_query failed: {rc}"
        return res

    def query_groups(self, plan, nthreads=1, cap=1 << 21):
        """Flat result for group counts beyond the sn_result page: returns
        the full sorted [(keys tuple, vals list), ...] like result_rows()."""
        keys = np.zeros((cap, SN_MAX_GROUPS, SN_KEY_MAX), dtype=np.uint8)
        knull = np.zeros((cap, SN_MAX_GROUPS), dtype=np.uint8)
        vals = np.zeros((cap, SN_MAX_AGGS), dtype=np.float64)
        vnull = np.zeros((cap, SN_MAX_AGGS), dtype=np.uint8)
        n = lib().sno_query_groups(
            self._h, C.byref(plan), nthreads, cap,
            keys.ctypes.data_as(C.c_char_p),
            knull.ctypes.data_as(C.POINTER(C.c_uint8)),
            vals.ctypes.data_as(C.POINTER(C.c_double)),
            vnull.ctypes.data_as(C.POINTER(C.c_uint8)))
        assert n >= 0, f"sno_query_groups failed: {n}"
        assert n <= cap, f"{n} groups exceed cap {cap}"
        ngroup = plan.ngroup if plan.join_dim < 0 or plan.join_mode != 1 else 1 + plan.ngroup
        naggs = plan.naggs
        out = []
        for i in range(n):
            ks = tuple(
                None if knull[i][k] else
                bytes(keys[i][k]).split(b"\0")[0].decode()
                for k in range(ngroup))
            vs = [None if vnull[i][a] else float(vals[i][a])
                  for a in range(naggs)]
            out.append((ks, vs))
        return out

    def __del__(self):
        try:
            if self._h:
                lib().sno_table_destroy(self._h)
        except Exception:
            pass


def make_plan(preds=(), group_cols=(), aggs=(), join=None):
    """preds: list of dicts {col, lo, hi, lo_strict, hi_strict, is_double}
    aggs: list of ('sum'|'avg'|'count', [(col, add, mul), ...])"""
    p = SnPlan()
    p.table = 0
    p.npreds = len(preds)
    for i, pr in enumerate(preds):
        sp = p.preds[i]
        sp.col = pr["col"]
        if pr.get("is_double"):
            if "lo" in pr:
                sp.lo_d = float(pr["lo"]); sp.has_lo = 1
            if "hi" in pr:
                sp.hi_d = float(pr["hi"]); sp.has_hi = 1
        else:
            if "lo" in pr:
                sp.lo_i = int(pr["lo"]); sp.has_lo = 1
            if "hi" in pr:
                sp.hi_i = int(pr["hi"]); sp.has_hi = 1
        sp.lo_strict = 1 if pr.get("lo_strict") else 0
        sp.hi_strict = 1 if pr.get("hi_strict") else 0
        if "eq" in pr:
            lit = pr["eq"] if isinstance(pr["eq"], bytes) else pr["eq"].encode()
            sp.str_eq = lit          # ctypes keeps the bytes alive via _objects
            sp.str_len = len(lit)
        if "in" in pr:
            vals = pr["in"]
            sp.in_n = len(vals)
            if vals and isinstance(vals[0], (bytes, str)):
                lits = [v if isinstance(v, bytes) else v.encode() for v in vals]
                arr = (C.c_char_p * len(lits))(*lits)
                lens = (C.c_int32 * len(lits))(*[len(v) for v in lits])
                sp.in_s = arr
                sp.in_s_len = lens
                p._keep = getattr(p, "_keep", []) + [arr, lens, lits]
            else:
                arr = (C.c_int64 * len(vals))(*[int(v) for v in vals])
                sp.in_i = arr
                p._keep = getattr(p, "_keep", []) + [arr]
    p.ngroup = len(group_cols)
    for i, c in enumerate(group_cols):
        p.group_cols[i] = c
    p.join_dim = -1
    if join is not None:
        p.join_dim = join["dim"]
        p.join_fact_col = join["fact_col"]
        p.join_mode = 1 if join.get("group") else 0
    p.naggs = len(aggs)
    for i, (kind, factors) in enumerate(aggs):
        ag = p.aggs[i]
        ag.kind = {"sum": AGG_SUM, "count": AGG_COUNT_STAR,
                   "avg": AGG_AVG, "min": AGG_MIN, "max": AGG_MAX}[kind]
        ag.nfactors = len(factors)
        for j, (col, add, mul) in enumerate(factors):
            ag.factors[j].col = col
            ag.factors[j].add = add
            ag.factors[j].mul = mul
    return p


def result_rows(res):
    """Returns list of (keys tuple, vals list) from SnResult."""
    out = []
    for r in range(res.nrows):
        keys = []
        for k in range(res.ngroup):
            if res.key_is_null[r][k]:
                keys.append(None)
            else:
                keys.append(bytes(res.keys[r][k].value).decode())
        vals = []
        for a in range(res.naggs):
            vals.append(None if res.val_is_null[r][a] else res.vals[r][a])
        out.append((tuple(keys), vals))
    return out
