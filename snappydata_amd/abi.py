"""ctypes mirror of include/snappy_engine.h (the engine's C ABI).

Product-side only — the host layer a JVM would reach over JNI is exactly this
surface; Python here is the test/bench harness standing in for that host.
"""
import ctypes as C

SN_MAX_PREDS = 8
SN_MAX_AGGS = 12
SN_MAX_GROUPS = 2
SN_MAX_FACTORS = 3
SN_MAX_GROUP_SLOTS = 1024
SN_KEY_MAX = 48

T_INT32, T_INT64, T_DOUBLE, T_STRING, T_BOOL, T_INT16, T_INT8, T_FLOAT = range(8)
AGG_SUM, AGG_COUNT_STAR, AGG_AVG, AGG_MIN, AGG_MAX = range(5)

OK = 0
ERR_NOGPU = -6


class SnBuf(C.Structure):
    _fields_ = [("data", C.c_void_p), ("len", C.c_int64)]


class SnColSchema(C.Structure):
    _fields_ = [("dtype", C.c_int32), ("nullable", C.c_int32)]


class SnConfig(C.Structure):
    _fields_ = [("device", C.c_int32),
                ("column_batch_size", C.c_int64),
                ("column_max_delta_rows", C.c_int32),
                ("hash_join_size", C.c_int64),
                ("n_buckets", C.c_int32),
                ("shard_rank", C.c_int32),
                ("shard_count", C.c_int32)]


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


class SnIngestCol(C.Structure):
    _fields_ = [("data", C.c_void_p), ("str_lens", C.POINTER(C.c_int32)),
                ("valid", C.POINTER(C.c_uint8))]


def make_plan(table=0, preds=(), group_cols=(), aggs=(), join=None):
    """Same plan-construction convention as the reference's thin planner
    (SnappyStrategies shapes): preds = [{col, lo, hi, lo_strict, hi_strict,
    is_double}], aggs = [(kind, [(col, add, mul), ...])]."""
    p = SnPlan()
    p.table = table
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
    out = []
    for r in range(res.nrows):
        keys = []
        for k in range(res.ngroup):
            keys.append(None if res.key_is_null[r][k]
                        else bytes(res.keys[r][k].value).decode())
        vals = [None if res.val_is_null[r][a] else res.vals[r][a]
                for a in range(res.naggs)]
        out.append((tuple(keys), vals))
    return out
