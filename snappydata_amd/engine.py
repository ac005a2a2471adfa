"""Python host layer over the MI355X engine (libsnappy_engine.so).

Mirrors the reference's operator/data surfaces a drop-in must honor:
  - Engine.table_define  -> ColumnFormatRelation metadata
  - Engine.batch_put     -> ExternalStore.storeColumnBatch
                            (ExternalStore.scala:43-45) / the
                            ColumnBatchIterator buffer protocol
                            (ColumnBatchIterator.scala:96-163)
  - Engine.query         -> ColumnTableScan -> Filter ->
                            SnappyHashAggregateExec execution
  - Query.partials/merge -> the partial->final exchange (ShuffleExchange
                            between SnappyHashAggregateExec stages)

The compute path is the HIP engine; this layer is plumbing.  There is NO CPU
fallback: querying without a GPU raises EngineError(SN_ERR_NOGPU).
"""
import ctypes as C
import os
import subprocess

import numpy as np

from . import abi

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libsnappy_engine.so")


class EngineError(RuntimeError):
    def __init__(self, code, msg=""):
        super().__init__(f"engine error {code}: {msg}")
        self.code = code


def build(force=False):
    """Compile the engine .so in-tree (hipcc --offload-arch=gfx950)."""
    srcdir = os.path.join(_DIR, "csrc")
    srcs = [os.path.join(srcdir, f) for f in
            ("engine.cpp", "builder.cpp", "kernels.hip", "engine_internal.h")]
    if force or (not os.path.exists(_SO)) or \
            os.path.getmtime(_SO) < max(os.path.getmtime(s) for s in srcs):
        subprocess.run(["make", "-C", srcdir], check=True)
    return _SO


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO):
            build()
        L = C.CDLL(_SO)
        L.sn_engine_create.restype = C.c_void_p
        L.sn_engine_create.argtypes = [C.POINTER(abi.SnConfig)]
        L.sn_engine_destroy.argtypes = [C.c_void_p]
        L.sn_last_error.restype = C.c_char_p
        L.sn_engine_arch.restype = C.c_char_p
        L.sn_table_define.restype = C.c_int32
        L.sn_table_define.argtypes = [C.c_void_p, C.c_char_p, C.c_int32,
                                      C.POINTER(abi.SnColSchema)]
        L.sn_batch_put.restype = C.c_int32
        L.sn_batch_put.argtypes = [C.c_void_p, C.c_int32, C.c_int64, C.c_int32,
                                   C.c_int32, C.POINTER(abi.SnBuf),
                                   C.POINTER(abi.SnBuf), C.POINTER(abi.SnBuf),
                                   C.POINTER(abi.SnBuf)]
        L.sn_table_num_batches.restype = C.c_int64
        L.sn_table_num_batches.argtypes = [C.c_void_p, C.c_int32]
        L.sn_table_num_rows.restype = C.c_int64
        L.sn_table_num_rows.argtypes = [C.c_void_p, C.c_int32]
        L.sn_table_get_blob.restype = C.c_int64
        L.sn_table_get_blob.argtypes = [C.c_void_p, C.c_int32, C.c_int32,
                                        C.c_int32, C.c_void_p, C.c_int64]
        L.sn_query_submit.restype = C.c_void_p
        L.sn_query_submit.argtypes = [C.c_void_p, C.POINTER(abi.SnPlan)]
        L.sn_query_wait.restype = C.c_int32
        L.sn_query_wait.argtypes = [C.c_void_p]
        L.sn_query_result.restype = C.c_int32
        L.sn_query_result.argtypes = [C.c_void_p, C.POINTER(abi.SnResult)]
        L.sn_query_destroy.argtypes = [C.c_void_p]
        L.sn_query_kernel_ms.restype = C.c_double
        L.sn_query_kernel_ms.argtypes = [C.c_void_p]
        L.sn_query_used_jit.restype = C.c_int32
        L.sn_query_used_jit.argtypes = [C.c_void_p]
        L.sn_query_order_by.restype = C.c_int32
        L.sn_query_order_by.argtypes = [C.c_void_p, C.c_int32, C.c_int32,
                                        C.c_int64]
        L.sn_query_result_page.restype = C.c_int32
        L.sn_query_result_page.argtypes = [C.c_void_p, C.c_int64,
                                           C.POINTER(abi.SnResult)]
        L.sn_query_num_groups.restype = C.c_int64
        L.sn_query_num_groups.argtypes = [C.c_void_p]
        L.sn_engine_jit_count.restype = C.c_int32
        L.sn_engine_jit_count.argtypes = [C.c_void_p]
        L.sn_query_partial_bytes.restype = C.c_int64
        L.sn_query_partial_bytes.argtypes = [C.c_void_p]
        L.sn_query_partials.restype = C.c_int32
        L.sn_query_partials.argtypes = [C.c_void_p, C.c_void_p, C.c_int32]
        L.sn_query_merge.restype = C.c_int32
        L.sn_query_merge.argtypes = [C.c_void_p, C.c_void_p, C.c_int64, C.c_int32]
        L.sn_query_partials_sharded.restype = C.c_int32
        L.sn_query_partials_sharded.argtypes = [C.c_void_p, C.c_int32, C.c_void_p]
        L.sn_query_partial_bytes2.restype = C.c_int64
        L.sn_query_partial_bytes2.argtypes = [C.c_void_p, C.c_int32]
        L.sn_query_partials2.restype = C.c_int32
        L.sn_query_partials2.argtypes = [C.c_void_p, C.c_void_p, C.c_int32,
                                         C.c_int32]
        L.sn_query_partials_sharded2.restype = C.c_int32
        L.sn_query_partials_sharded2.argtypes = [C.c_void_p, C.c_int32,
                                                 C.c_void_p, C.c_int32]
        L.sn_batch_mutate.restype = C.c_int32
        L.sn_batch_mutate.argtypes = [C.c_void_p, C.c_int32, C.c_int64,
                                      C.c_int32, C.POINTER(abi.SnBuf),
                                      C.POINTER(abi.SnBuf), C.POINTER(abi.SnBuf)]
        L.sn_ingest_columns.restype = C.c_int64
        L.sn_ingest_columns.argtypes = [C.c_void_p, C.c_int32, C.c_int64,
                                        C.POINTER(abi.SnIngestCol), C.c_int32,
                                        C.c_int32]
        L.sn_datagen_lineitem.restype = C.c_int64
        L.sn_datagen_lineitem.argtypes = [C.c_void_p, C.c_int32, C.c_int64,
                                          C.c_int64, C.c_int32, C.c_int32]
        L.sn_dim_define.restype = C.c_int32
        L.sn_dim_define.argtypes = [C.c_void_p, C.c_char_p]
        L.sn_dim_put.restype = C.c_int32
        L.sn_dim_put.argtypes = [C.c_void_p, C.c_int32, C.c_int64,
                                 C.POINTER(C.c_int64), C.c_char_p,
                                 C.POINTER(C.c_int32)]
        L.sn_dim_from_table.restype = C.c_int32
        L.sn_dim_from_table.argtypes = [C.c_void_p, C.c_int32, C.c_int32,
                                        C.c_int32, C.c_int32]
        L.sn_encode_column.restype = C.c_int64
        L.sn_encode_column.argtypes = [C.c_int32, C.c_void_p, C.POINTER(C.c_int32),
                                       C.POINTER(C.c_uint8), C.c_int32,
                                       C.c_void_p, C.c_int64]
        L.sn_encode_delete_mask.restype = C.c_int64
        L.sn_encode_delete_mask.argtypes = [C.POINTER(C.c_int32), C.c_int32,
                                            C.c_int32, C.c_void_p, C.c_int64]
        L.sn_encode_update_delta.restype = C.c_int64
        L.sn_encode_update_delta.argtypes = [C.c_int32, C.POINTER(C.c_int32),
                                             C.c_int32, C.c_int32, C.c_void_p,
                                             C.POINTER(C.c_int32),
                                             C.POINTER(C.c_uint8), C.c_void_p,
                                             C.c_int64]
        L.sn_gen_lineitem_arrays.argtypes = [
            C.c_int64, C.c_int32, C.c_int64,
            C.POINTER(C.c_double), C.POINTER(C.c_double), C.POINTER(C.c_double),
            C.POINTER(C.c_double), C.POINTER(C.c_uint8), C.POINTER(C.c_uint8),
            C.POINTER(C.c_int32)]
        _lib = L
    return _lib


def last_error():
    return (lib().sn_last_error() or b"").decode()


def _check(rc, what=""):
    if rc < 0:
        raise EngineError(rc, f"{what}: {last_error()}")
    return rc


def encode_column(dtype, data, lens=None, valid=None):
    """Product column encoder (reference blob format)."""
    data = np.ascontiguousarray(data) if not isinstance(data, (bytes, bytearray)) \
        else np.frombuffer(data, dtype=np.uint8)
    count = len(lens) if lens is not None else len(data)
    cap = 64 + int(data.nbytes) * 2 + count * 8
    out = np.zeros(cap, dtype=np.uint8)
    lp = np.ascontiguousarray(lens, dtype=np.int32).ctypes.data_as(
        C.POINTER(C.c_int32)) if lens is not None else None
    vp = np.ascontiguousarray(valid, dtype=np.uint8).ctypes.data_as(
        C.POINTER(C.c_uint8)) if valid is not None else None
    n = _check(lib().sn_encode_column(dtype, data.ctypes.data, lp, vp, count,
                                      out.ctypes.data, cap), "encode_column")
    return out[:n].tobytes()


def encode_delete_mask(positions, num_base_rows):
    pos = np.ascontiguousarray(positions, dtype=np.int32)
    cap = 16 + 4 * len(pos)
    out = np.zeros(cap, dtype=np.uint8)
    n = _check(lib().sn_encode_delete_mask(
        pos.ctypes.data_as(C.POINTER(C.c_int32)), len(pos), num_base_rows,
        out.ctypes.data, cap), "encode_delete_mask")
    return out[:n].tobytes()


def encode_update_delta(dtype, positions, num_base_rows, values, valid=None):
    pos = np.ascontiguousarray(positions, dtype=np.int32)
    vals = np.ascontiguousarray(values)
    cap = 128 + len(pos) * 40 + int(vals.nbytes) * 2
    out = np.zeros(cap, dtype=np.uint8)
    vp = np.ascontiguousarray(valid, dtype=np.uint8).ctypes.data_as(
        C.POINTER(C.c_uint8)) if valid is not None else None
    n = _check(lib().sn_encode_update_delta(
        dtype, pos.ctypes.data_as(C.POINTER(C.c_int32)), len(pos),
        num_base_rows, vals.ctypes.data, None, vp, out.ctypes.data, cap),
        "encode_update_delta")
    return out[:n].tobytes()


def gen_lineitem_arrays(start_row, n, seed):
    """Generate the synthetic lineitem columns (same generator as the engine's
    sn_datagen_lineitem — used by tests and bench's CPU-baseline sample)."""
    qty = np.empty(n); ep = np.empty(n); disc = np.empty(n); tax = np.empty(n)
    rf = np.empty(n, dtype=np.uint8); ls = np.empty(n, dtype=np.uint8)
    ship = np.empty(n, dtype=np.int32)
    lib().sn_gen_lineitem_arrays(
        start_row, n, seed,
        qty.ctypes.data_as(C.POINTER(C.c_double)),
        ep.ctypes.data_as(C.POINTER(C.c_double)),
        disc.ctypes.data_as(C.POINTER(C.c_double)),
        tax.ctypes.data_as(C.POINTER(C.c_double)),
        rf.ctypes.data_as(C.POINTER(C.c_uint8)),
        ls.ctypes.data_as(C.POINTER(C.c_uint8)),
        ship.ctypes.data_as(C.POINTER(C.c_int32)))
    return dict(qty=qty, ep=ep, disc=disc, tax=tax,
                rf=[bytes([b]) for b in rf], ls=[bytes([b]) for b in ls],
                ship=ship)


class Query:
    def __init__(self, eng, handle, plan):
        self._e = eng
        self._h = handle
        self.plan = plan

    def wait(self):
        _check(lib().sn_query_wait(self._h), "query_wait")
        return self

    def result(self):
        res = abi.SnResult()
        _check(lib().sn_query_result(self._h, C.byref(res)), "query_result")
        return res

    def rows(self):
        total = _check(lib().sn_query_num_groups(self._h), "num_groups")
        if total <= abi.SN_MAX_GROUP_SLOTS:
            return abi.result_rows(self.result())
        out = []
        off = 0
        while off < total:
            res = abi.SnResult()
            _check(lib().sn_query_result_page(self._h, off, C.byref(res)),
                   "result_page")
            if res.nrows == 0:
                break
            out.extend(abi.result_rows(res))
            off += res.nrows
        return out

    def kernel_ms(self):
        """Scan-kernel duration (HIP events on the launch stream)."""
        return lib().sn_query_kernel_ms(self._h)

    def used_jit(self):
        """True when the query ran a query-compiled (hipRTC) kernel."""
        return bool(lib().sn_query_used_jit(self._h))

    def order_by(self, agg_idx, descending=False, k=0):
        """ORDER BY <aggregate value> [DESC] LIMIT k epilogue
        (SnappySortExec / TakeOrderedAndProject semantics)."""
        _check(lib().sn_query_order_by(self._h, agg_idx,
                                       1 if descending else 0, k), "order_by")
        return self

    def num_groups(self):
        """Total group rows of the local (pre-merge) result."""
        return _check(lib().sn_query_num_groups(self._h), "num_groups")

    def partial_bytes(self, cap=None):
        """Partial-block size; cap overrides the slot capacity for group
        counts beyond SN_MAX_GROUP_SLOTS (negotiate one cap across ranks)."""
        if cap is None:
            return _check(lib().sn_query_partial_bytes(self._h))
        return _check(lib().sn_query_partial_bytes2(self._h, cap))

    def partials_host(self, cap=None):
        n = self.partial_bytes(cap)
        buf = np.zeros(n, dtype=np.uint8)
        if cap is None:
            _check(lib().sn_query_partials(self._h, buf.ctypes.data, 0),
                   "partials")
        else:
            _check(lib().sn_query_partials2(self._h, buf.ctypes.data, 0, cap),
                   "partials")
        return buf

    def partials_sharded(self, world, cap=None):
        """Key-sharded split for the grouped all-to-all:
        returns (world, partial_bytes) uint8; row d travels to rank d."""
        bb = self.partial_bytes(cap)
        buf = np.zeros(world * bb, dtype=np.uint8)
        if cap is None:
            _check(lib().sn_query_partials_sharded(self._h, world,
                                                   buf.ctypes.data),
                   "partials_sharded")
        else:
            _check(lib().sn_query_partials_sharded2(self._h, world,
                                                    buf.ctypes.data, cap),
                   "partials_sharded")
        return buf.reshape(world, bb)

    def partials_into_device(self, dev_ptr):
        _check(lib().sn_query_partials(self._h, C.c_void_p(dev_ptr), 1), "partials")

    def merge_host(self, blocks, stride, n_blocks):
        """blocks: numpy uint8 array of n_blocks partial blocks."""
        _check(lib().sn_query_merge(self._h, blocks.ctypes.data, stride, n_blocks),
               "merge")
        return self

    def close(self):
        if self._h:
            lib().sn_query_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class Engine:
    def __init__(self, device=0, shard_rank=0, shard_count=1, n_buckets=0,
                 column_batch_size=0):
        cfg = abi.SnConfig()
        cfg.device = device
        cfg.shard_rank = shard_rank
        cfg.shard_count = shard_count
        cfg.n_buckets = n_buckets
        cfg.column_batch_size = column_batch_size
        self._h = lib().sn_engine_create(C.byref(cfg))
        if not self._h:
            raise EngineError(-1, last_error())

    def table_define(self, name, schema):
        """schema: list of (dtype, nullable)."""
        arr = (abi.SnColSchema * len(schema))()
        for i, (dt, nl) in enumerate(schema):
            arr[i].dtype = dt
            arr[i].nullable = 1 if nl else 0
        return _check(lib().sn_table_define(self._h, name.encode(), len(schema), arr),
                      "table_define")

    def batch_put(self, table, uuid, bucket, num_rows, col_blobs,
                  stats=None, delete_mask=None, deltas=None):
        nc = len(col_blobs)
        bufs = (abi.SnBuf * nc)()
        keep = []
        for i, blob in enumerate(col_blobs):
            a = np.frombuffer(blob, dtype=np.uint8)
            keep.append(a)
            bufs[i].data = a.ctypes.data
            bufs[i].len = len(blob)
        sb = abi.SnBuf()
        if stats:
            a = np.frombuffer(stats, dtype=np.uint8); keep.append(a)
            sb.data, sb.len = a.ctypes.data, len(stats)
        db = abi.SnBuf()
        if delete_mask:
            a = np.frombuffer(delete_mask, dtype=np.uint8); keep.append(a)
            db.data, db.len = a.ctypes.data, len(delete_mask)
        dl = None
        if deltas is not None:
            dl = (abi.SnBuf * (nc * 2))()
            for i, pair in enumerate(deltas):
                for j, d in enumerate(pair):
                    if d:
                        a = np.frombuffer(d, dtype=np.uint8); keep.append(a)
                        dl[i * 2 + j].data, dl[i * 2 + j].len = a.ctypes.data, len(d)
        _check(lib().sn_batch_put(self._h, table, uuid, bucket, num_rows, bufs,
                                  C.byref(sb) if stats else None,
                                  C.byref(db) if delete_mask else None, dl),
               "batch_put")

    def num_batches(self, table):
        return _check(lib().sn_table_num_batches(self._h, table))

    def num_rows(self, table):
        return _check(lib().sn_table_num_rows(self._h, table))

    def get_blob(self, table, batch, col, cap=1 << 26):
        buf = np.zeros(cap, dtype=np.uint8)
        n = _check(lib().sn_table_get_blob(self._h, table, batch, col,
                                           buf.ctypes.data, cap), "get_blob")
        return bytes(buf[:n].tobytes())

    def ingest_columns(self, table, cols, nrows, batch_rows=200000, first_bucket=0):
        """cols: list of dicts {data: np.ndarray|bytes payload, lens: np.int32
        array (strings), valid: np.uint8 array or None}."""
        arr = (abi.SnIngestCol * len(cols))()
        keep = []
        for i, col in enumerate(cols):
            data = col["data"]
            if isinstance(data, (bytes, bytearray)):
                data = np.frombuffer(data, dtype=np.uint8)
            data = np.ascontiguousarray(data)
            keep.append(data)
            arr[i].data = data.ctypes.data
            if col.get("lens") is not None:
                lens = np.ascontiguousarray(col["lens"], dtype=np.int32)
                keep.append(lens)
                arr[i].str_lens = lens.ctypes.data_as(C.POINTER(C.c_int32))
            if col.get("valid") is not None:
                v = np.ascontiguousarray(col["valid"], dtype=np.uint8)
                keep.append(v)
                arr[i].valid = v.ctypes.data_as(C.POINTER(C.c_uint8))
        return _check(lib().sn_ingest_columns(self._h, table, nrows, arr,
                                              batch_rows, first_bucket), "ingest")

    def datagen_lineitem(self, table, total_rows, seed=42, batch_rows=0, threads=0):
        return _check(lib().sn_datagen_lineitem(self._h, table, total_rows, seed,
                                                batch_rows, threads), "datagen")

    def jit_count(self):
        """Compiled kernels in the tokenized plan cache (shapes, not
        literal values)."""
        return lib().sn_engine_jit_count(self._h)

    def batch_mutate(self, table, uuid, bucket, delete_mask=None,
                     deltas=None, stats=None):
        """Attach the CURRENT cumulative mutation state to an existing batch
        (the UPDATE/DELETE seam): delete_mask/deltas replace prior state."""
        keep = []

        def buf(blob):
            b = abi.SnBuf()
            if blob:
                a = np.frombuffer(blob, dtype=np.uint8)
                keep.append(a)
                b.data = a.ctypes.data
                b.len = len(blob)
            return b

        dm = buf(delete_mask) if delete_mask else None
        st = buf(stats) if stats else None
        darr = None
        if deltas is not None:
            nc = len(deltas)
            darr = (abi.SnBuf * (2 * nc))()
            for i, pair in enumerate(deltas):
                d1, d2 = pair if isinstance(pair, tuple) else (pair, None)
                darr[2 * i] = buf(d1)
                darr[2 * i + 1] = buf(d2)
        _check(lib().sn_batch_mutate(
            self._h, table, uuid, bucket,
            C.byref(dm) if dm else None, darr,
            C.byref(st) if st else None), "batch_mutate")

    def dim_define(self, name):
        return _check(lib().sn_dim_define(self._h, name.encode()), "dim_define")

    def dim_put(self, dim, keys, attrs=None):
        """Load dimension rows (the row-store stand-in): keys int64, attrs
        optional list of bytes per key (for group-by-dim-attr joins)."""
        keys = np.ascontiguousarray(keys, dtype=np.int64)
        payload, lens = None, None
        if attrs is not None:
            lens = np.array([len(a) for a in attrs], dtype=np.int32)
            payload = b"".join(attrs)
        _check(lib().sn_dim_put(
            self._h, dim, len(keys),
            keys.ctypes.data_as(C.POINTER(C.c_int64)), payload,
            lens.ctypes.data_as(C.POINTER(C.c_int32)) if lens is not None else None),
            "dim_put")

    def dim_from_table(self, dim, table, key_col, attr_col=-1):
        """Populate an empty dimension from a resident column table (the
        colocated partitioned-partitioned join build, device-side —
        HashJoinExec per-task build + HashedObjectCache reuse)."""
        _check(lib().sn_dim_from_table(self._h, dim, table, key_col, attr_col),
               "dim_from_table")

    def query(self, plan):
        h = lib().sn_query_submit(self._h, C.byref(plan))
        if not h:
            raise EngineError(-1, last_error())
        return Query(self, h, plan)

    def close(self):
        if self._h:
            lib().sn_engine_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
