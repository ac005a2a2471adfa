"""CPU-side randomized cross-check of the ORACLE itself: random schemas,
encodings, deletes, update deltas and plans evaluated BOTH through the
oracle (the reference-loop restatement all GPU parity anchors on) and an
independent row-at-a-time numpy evaluation written directly from Spark's
aggregate semantics.  Triangulates the parity anchor — a bug common to
the oracle and the engine would otherwise be invisible to the GPU fuzz.

Runs in the CPU suite every round (SN_ORACLE_FUZZ_N cases, default 12).
"""
import os

import numpy as np

from oracle import pyoracle as po

REL = 1e-9
VOCAB = [b"AA", b"BEE", b"CEE", b"DEE", b"EFF", b"GEE"]
T_NUM = [po.T_INT32, po.T_INT64, po.T_DOUBLE, po.T_FLOAT, po.T_INT16]


def _gen_numeric(rng, dtype, n):
    if dtype == po.T_DOUBLE:
        return rng.random(n) * 200 - 100
    if dtype == po.T_FLOAT:
        return (rng.random(n) * 50).astype(np.float32)
    if dtype == po.T_INT64:
        return rng.integers(-(1 << 40), 1 << 40, n)
    if dtype == po.T_INT32:
        return rng.integers(-5_000, 5_000, n).astype(np.int32)
    return rng.integers(-300, 300, n).astype(np.int16)


def _case(rng):
    ncols = int(rng.integers(2, 5))
    schema = []
    for c in range(ncols):
        if c > 0 and rng.random() < 0.3:
            schema.append((po.T_STRING, bool(rng.random() < 0.3)))
        else:
            schema.append((T_NUM[rng.integers(0, len(T_NUM))],
                           bool(rng.random() < 0.3)))
    batches = []
    for _ in range(int(rng.integers(1, 4))):
        n = int(rng.integers(500, 8_000))
        cols = []
        for d, nullable in schema:
            if d == po.T_STRING:
                v = [VOCAB[i] for i in rng.integers(0, len(VOCAB), n)]
                if nullable:
                    for i in np.flatnonzero(rng.random(n) < 0.15):
                        v[i] = None
                cols.append((v, None))
            else:
                v = _gen_numeric(rng, d, n)
                valid = ((rng.random(n) >= 0.15).astype(np.uint8)
                         if nullable else None)
                cols.append((v, valid))
        dels = (np.unique(rng.integers(0, n, n // 15)).astype(np.int32)
                if rng.random() < 0.4 else None)
        # update deltas (2-deep) on one numeric column
        deltas = None
        numc = [c for c in range(ncols) if schema[c][0] != po.T_STRING]
        if numc and rng.random() < 0.35:
            dc = int(rng.choice(numc))
            deltas = {"col": dc, "layers": []}
            for _d in range(1 if rng.random() < 0.5 else 2):
                pos = np.unique(rng.integers(0, n, max(1, n // 30))).astype(np.int32)
                deltas["layers"].append(
                    (pos, _gen_numeric(rng, schema[dc][0], len(pos))))
        batches.append((n, cols, dels, deltas))
    # plan
    numc = [c for c in range(ncols) if schema[c][0] != po.T_STRING]
    strc = [c for c in range(ncols) if schema[c][0] == po.T_STRING]
    preds = []
    for c in rng.permutation(numc)[: rng.integers(0, 3)]:
        d = schema[c][0]
        is_d = d in (po.T_DOUBLE, po.T_FLOAT)
        p = dict(col=int(c))
        if is_d:
            lo, hi = sorted(rng.random(2) * 200 - 100)
            p["is_double"] = True
        else:
            lo, hi = sorted(rng.integers(-4_000, 4_000, 2).tolist())
            lo, hi = int(lo), int(hi)
        if rng.random() < 0.8:
            p["lo"] = float(lo) if is_d else lo
            p["lo_strict"] = bool(rng.random() < 0.3)
        if rng.random() < 0.8:
            p["hi"] = float(hi) if is_d else hi
            p["hi_strict"] = bool(rng.random() < 0.3)
        if "lo" in p or "hi" in p:
            preds.append(p)
    if strc and rng.random() < 0.4:
        picks = [VOCAB[i] for i in rng.choice(len(VOCAB), 2, replace=False)]
        preds.append({"col": int(strc[0]), "in": picks})
    aggs = []
    for _ in range(int(rng.integers(1, 4))):
        if not numc:
            break
        kind = ["sum", "avg", "min", "max"][rng.integers(0, 4)]
        nf = 2 if len(numc) > 1 and rng.random() < 0.3 else 1
        fac = [(int(fc), float(rng.integers(-2, 3)),
                float(rng.choice([1.0, 1.0, 0.5, 2.0])))
               for fc in rng.choice(numc, nf, replace=False)]
        aggs.append((kind, fac))
    aggs.append(("count", []))
    group_cols = []
    if rng.random() < 0.6:
        cands = strc + [c for c in numc
                        if schema[c][0] in (po.T_INT16, po.T_INT32, po.T_INT64)]
        if cands:
            group_cols = [int(rng.choice(cands))]
    return schema, batches, dict(preds=preds, group_cols=group_cols, aggs=aggs)


def _np_eval(schema, batches, plan_kw):
    """Independent row-level reference (Spark semantics: predicates drop
    null operands, Sum/Avg/Min/Max skip rows with any null factor)."""
    groups = {}
    order = []
    if not plan_kw["group_cols"]:
        groups[()] = [[0.0, 0.0, None] for _ in plan_kw["aggs"]]
        order.append(())          # keyless always yields one row
    for n, cols, dels, deltas in batches:
        vals = []
        for c, ((v, valid), (d, nullable)) in enumerate(zip(cols, schema)):
            if d == po.T_STRING:
                vv = list(v)
            else:
                vv = [None if (valid is not None and not valid[i])
                      else float(v[i]) for i in range(n)]
            vals.append(vv)
        if deltas is not None:
            dc = deltas["col"]
            # layers appended oldest-first in generation order; the LAST
            # generated layer is delta1 (newest, wins)
            for pos, pv in deltas["layers"]:
                for i, p_ in enumerate(pos):
                    vals[dc][p_] = float(pv[i])
        keep = np.ones(n, dtype=bool)
        if dels is not None:
            keep[dels] = False
        for r in range(n):
            if not keep[r]:
                continue
            ok = True
            for p in plan_kw["preds"]:
                x = vals[p["col"]][r]
                if x is None:
                    ok = False
                    break
                if "in" in p:
                    if x not in p["in"]:
                        ok = False
                        break
                    continue
                if "lo" in p and (x < p["lo"] or
                                  (p.get("lo_strict") and x == p["lo"])):
                    ok = False
                    break
                if "hi" in p and (x > p["hi"] or
                                  (p.get("hi_strict") and x == p["hi"])):
                    ok = False
                    break
            if not ok:
                continue
            if plan_kw["group_cols"]:
                gc = plan_kw["group_cols"][0]
                kv = vals[gc][r]
                if kv is None:
                    key = (None,)
                elif isinstance(kv, bytes):
                    key = (kv.decode(),)
                else:
                    key = (str(int(kv)),)
            else:
                key = ()
            if key not in groups:
                groups[key] = [[0.0, 0.0, None] for _ in plan_kw["aggs"]]
                order.append(key)
            g = groups[key]
            for a, (kind, fac) in enumerate(plan_kw["aggs"]):
                if kind == "count":
                    g[a][0] += 1
                    g[a][1] += 1
                    continue
                prod, anynull = 1.0, False
                for (fc, add, mul) in fac:
                    x = vals[fc][r]
                    if x is None:
                        anynull = True
                        break
                    prod *= add + mul * x
                if anynull:
                    continue
                g[a][1] += 1
                if kind in ("sum", "avg"):
                    g[a][0] += prod
                else:
                    cur = g[a][2]
                    g[a][2] = (prod if cur is None else
                               (min(cur, prod) if kind == "min"
                                else max(cur, prod)))
    out = []
    for key in order:
        g = groups[key]
        row = []
        for a, (kind, fac) in enumerate(plan_kw["aggs"]):
            s, cnt, mm = g[a]
            if kind == "count":
                row.append(s)
            elif cnt == 0:
                row.append(None)
            elif kind == "sum":
                row.append(s)
            elif kind == "avg":
                row.append(s / cnt)
            else:
                row.append(mm)
        out.append((key, tuple(row)))
    return out


def test_oracle_vs_numpy_fuzz():
    n_cases = int(os.environ.get("SN_ORACLE_FUZZ_N", "12"))
    base = int(os.environ.get("SN_ORACLE_FUZZ_BASE", "7000"))
    ran = 0
    for seed in range(base, base + n_cases):
        rng = np.random.default_rng(seed)
        schema, batches, plan_kw = _case(rng)
        ot = po.OracleTable([d for d, _ in schema])
        for n, cols, dels, deltas in batches:
            blobs = []
            for c, ((v, valid), (d, _nb)) in enumerate(zip(cols, schema)):
                enc = po.ENC_DICT if d == po.T_STRING else po.ENC_UNCOMPRESSED
                blobs.append(po.encode(d, enc, v, valid=valid))
            dmask = po.encode_delete(dels, n) if dels is not None else None
            dd = None
            if deltas is not None:
                dd = [(None, None)] * len(schema)
                dc = deltas["col"]
                layers = deltas["layers"]
                blobs_l = [po.encode_delta(schema[dc][0], po.ENC_UNCOMPRESSED,
                                           pos, n, pv)
                           for pos, pv in layers]
                # newest layer (last generated) is delta1
                dd[dc] = ((blobs_l[1], blobs_l[0]) if len(blobs_l) == 2
                          else (blobs_l[0], None))
            ot.add_batch(-n if dd else n, blobs, delete_mask=dmask, deltas=dd)
        ref = _np_eval(schema, batches, plan_kw)
        plan = po.make_plan(**plan_kw)
        if plan_kw["group_cols"]:
            orows = ot.query_groups(plan, nthreads=4)
        else:
            orows = po.result_rows(ot.query(plan))
        km = {k: v for k, v in ref}
        assert len(orows) == len(ref), (seed, len(orows), len(ref))
        for k, ov in orows:
            rv = km[k]
            for a, (o, r) in enumerate(zip(ov, rv)):
                if r is None or o is None:
                    assert o == r, (seed, k, a, o, r)
                else:
                    assert abs(o - r) <= REL * max(1.0, abs(r)), (seed, k, a, o, r)
        ran += 1
    assert ran == n_cases
