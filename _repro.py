import ctypes, sys
import numpy as np
from snappydata_amd import engine as se
import tests.test_fuzz_parity as fz
hip = ctypes.CDLL("libamdhip64.so")
e = se.Engine(device=0)
for seed in range(1000, 1100):
    try:
        fz._run_case(e, seed)
        ok = "pass"
    except AssertionError as ex:
        ok = "FAIL " + str(ex)[:60]
    rc = hip.hipDeviceSynchronize()
    le = hip.hipGetLastError()
    if rc != 0 or le != 0 or ok != "pass":
        print(seed, ok, "sync_rc=", rc, "lastErr=", le, flush=True)
print("done")
