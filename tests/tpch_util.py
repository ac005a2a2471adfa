"""TPC-H helpers shared by tests and bench: lineitem schema, .tbl loading,
Q1/Q6 plan construction, golden expectations.

The engine-side lineitem projection (the 7 columns on the Q1/Q6 hot path):
  col 0 l_quantity      double
  col 1 l_extendedprice double
  col 2 l_discount      double
  col 3 l_tax           double
  col 4 l_returnflag    string (dictionary)
  col 5 l_linestatus    string (dictionary)
  col 6 l_shipdate      int32 (days since 1970-01-01, Spark DateType)

Types follow the reference DataFrame load path (TPCHTableSchema.scala:145-163:
measures double, dates DateType).  Query parameters follow
TPCH_Queries.scala:125-148 (Q1, DELTA=90) and :600-613 (Q6, DATE=1994-01-01,
DISCOUNT=0.06, QUANTITY=24), bounds computed in IEEE double exactly as
catalyst constant-folding would.
"""
import datetime
import os

from oracle import pyoracle as po

REFERENCE_TPCH = "/root/reference/tests/common/src/main/resources/TPCH"

LINEITEM_DTYPES = [po.T_DOUBLE, po.T_DOUBLE, po.T_DOUBLE, po.T_DOUBLE,
                   po.T_STRING, po.T_STRING, po.T_INT32]
COL_QTY, COL_EP, COL_DISC, COL_TAX, COL_RF, COL_LS, COL_SHIP = range(7)

EPOCH = datetime.date(1970, 1, 1)


def days(y, m, d):
    return (datetime.date(y, m, d) - EPOCH).days


def load_lineitem_tbl(path=None):
    """Parse the bundled lineitem.tbl into column lists (30,201 rows)."""
    import numpy as np
    path = path or os.path.join(REFERENCE_TPCH, "lineitem.tbl")
    qty, ep, disc, tax, rf, ls, ship = [], [], [], [], [], [], []
    with open(path, "rb") as f:
        for line in f:
            parts = line.rstrip(b"\n").split(b"|")
            if len(parts) < 16:
                continue
            qty.append(float(parts[4]))
            ep.append(float(parts[5]))
            disc.append(float(parts[6]))
            tax.append(float(parts[7]))
            rf.append(parts[8])
            ls.append(parts[9])
            y, m, d = parts[10].split(b"-")
            ship.append(days(int(y), int(m), int(d)))
    return {
        "qty": np.array(qty), "ep": np.array(ep), "disc": np.array(disc),
        "tax": np.array(tax), "rf": rf, "ls": ls,
        "ship": np.array(ship, dtype=np.int32),
    }


def encode_lineitem_batches(data, batch_rows=4096, with_stats=True):
    """Encode lineitem columns into reference-format batches.
    Returns list of (num_rows, col_blobs, stats)."""
    import numpy as np
    n = len(data["qty"])
    out = []
    for s in range(0, n, batch_rows):
        e = min(n, s + batch_rows)
        cols = [
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, data["qty"][s:e]),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, data["ep"][s:e]),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, data["disc"][s:e]),
            po.encode(po.T_DOUBLE, po.ENC_UNCOMPRESSED, data["tax"][s:e]),
            po.encode(po.T_STRING, po.ENC_DICT, data["rf"][s:e]),
            po.encode(po.T_STRING, po.ENC_DICT, data["ls"][s:e]),
            po.encode(po.T_INT32, po.ENC_UNCOMPRESSED, data["ship"][s:e]),
        ]
        stats = None
        if with_stats:
            lower = [float(np.min(data["qty"][s:e])), float(np.min(data["ep"][s:e])),
                     float(np.min(data["disc"][s:e])), float(np.min(data["tax"][s:e])),
                     0, 0, int(np.min(data["ship"][s:e]))]
            upper = [float(np.max(data["qty"][s:e])), float(np.max(data["ep"][s:e])),
                     float(np.max(data["disc"][s:e])), float(np.max(data["tax"][s:e])),
                     0, 0, int(np.max(data["ship"][s:e]))]
            stats = po.encode_stats(LINEITEM_DTYPES, e - s, lower, upper)
        out.append((e - s, cols, stats))
    return out


def q6_plan():
    """TPCH_Queries.scala:600-613 with DATE=1994-01-01, DISCOUNT=0.06,
    QUANTITY=24: revenue = sum(ep*disc) where shipdate in [1994-01-01,
    1995-01-01), discount between 0.06-0.01 and 0.06+0.01, quantity < 24.
    Catalyst parses the 0.06/0.01 literals as DECIMAL and folds them exactly,
    so the double-comparison bounds are the parsed doubles 0.05 and 0.07
    (verified against Snappy_6.out: IEEE-double folding would give
    0.049999999999999996/0.06999999999999999 and a different result)."""
    return po.make_plan(
        preds=[
            dict(col=COL_SHIP, lo=days(1994, 1, 1), hi=days(1995, 1, 1), hi_strict=True),
            dict(col=COL_DISC, is_double=True, lo=0.05, hi=0.07),
            dict(col=COL_QTY, is_double=True, hi=24.0, hi_strict=True),
        ],
        aggs=[("sum", [(COL_EP, 0.0, 1.0), (COL_DISC, 0.0, 1.0)])],
    )


def q1_plan():
    """TPCH_Queries.scala:125-148 with DELTA=90:
    shipdate <= date_sub('1997-12-31', 90) = 1997-10-02; group by
    returnflag, linestatus; 8 aggregates; order by keys."""
    cutoff = days(1997, 12, 31) - 90
    assert cutoff == days(1997, 10, 2)
    return po.make_plan(
        preds=[dict(col=COL_SHIP, hi=cutoff)],
        group_cols=[COL_RF, COL_LS],
        aggs=[
            ("sum", [(COL_QTY, 0.0, 1.0)]),
            ("sum", [(COL_EP, 0.0, 1.0)]),
            ("sum", [(COL_EP, 0.0, 1.0), (COL_DISC, 1.0, -1.0)]),
            ("sum", [(COL_EP, 0.0, 1.0), (COL_DISC, 1.0, -1.0), (COL_TAX, 1.0, 1.0)]),
            ("avg", [(COL_QTY, 0.0, 1.0)]),
            ("avg", [(COL_EP, 0.0, 1.0)]),
            ("avg", [(COL_DISC, 0.0, 1.0)]),
            ("count", []),
        ],
    )


def fmt(v):
    """The reference result formatting: '%18.4f'.format(d).trim() for doubles
    (QueryExecutor.scala:155-157); longs printed as-is."""
    return ("%18.4f" % v).strip()


def load_golden(qnum):
    p = os.path.join(REFERENCE_TPCH, "RESULT", f"Snappy_{qnum}.out")
    with open(p) as f:
        return [line.strip() for line in f if line.strip()]


# Committed copies of the reference's golden expectations for GPU-box runs
# (where /root/reference does not exist).  Source:
# tests/common/src/main/resources/TPCH/RESULT/Snappy_6.out, Snappy_1.out.
GOLDEN_Q6 = ["596503.1903"]
GOLDEN_Q1 = [
    "A,F,189203.0000,264917151.2300,251722566.7143,261813769.8429,25.2878,35407.2643,0.0501,7482",
    "N,F,4654.0000,6647990.5200,6333568.4966,6584905.2644,26.0000,37139.6118,0.0485,179",
    "N,O,269194.0000,376707514.9700,357845536.3767,372104781.3322,25.6302,35866.6586,0.0500,10503",
    "R,F,191214.0000,267924304.1400,254547618.0700,264804365.8424,25.6732,35972.6509,0.0498,7448",
]


def q1_result_lines(rows):
    """Format oracle/engine Q1 result rows the way the reference test does."""
    lines = []
    for keys, vals in rows:
        parts = list(keys)
        for i, v in enumerate(vals):
            if i == 7:  # count_order is a Long
                parts.append(str(int(v)))
            else:
                parts.append(fmt(v))
        lines.append(",".join(parts))
    return lines
