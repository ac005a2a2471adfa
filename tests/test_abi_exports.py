"""The C-ABI surface: every entry point include/snappy_engine.h declares must
load and resolve from the built library (no compute calls without a GPU)."""
import ctypes
import os
import re

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "snappy_engine.h")
SO = os.path.join(REPO, "snappydata_amd", "libsnappy_engine.so")


def declared_functions():
    src = open(HEADER).read()
    # strip comments
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)
    src = re.sub(r"//[^\n]*", "", src)
    return sorted(set(re.findall(r"\b(sn_[a-z0-9_]+)\s*\(", src)))


def test_header_declares_expected_surface():
    fns = declared_functions()
    for must in ["sn_engine_create", "sn_engine_destroy", "sn_table_define",
                 "sn_batch_put", "sn_query_submit", "sn_query_wait",
                 "sn_query_result", "sn_query_partials", "sn_query_merge",
                 "sn_last_error"]:
        assert must in fns, must


def test_all_declared_symbols_resolve():
    from snappydata_amd import engine as se
    se.build()
    lib = ctypes.CDLL(SO)
    for fn in declared_functions():
        assert getattr(lib, fn, None) is not None, f"missing export: {fn}"
