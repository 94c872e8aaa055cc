# The native C++ cpu_baseline leg (oracle/native/libhx_cpuref.so) must agree
# with the oracle on the workloads it is timed on (PLAIN pages, uncompressed
# or Snappy) — it is a timing denominator, but a wrong denominator is no
# denominator. CPU-only test (no GPU, no HIP).
import ctypes
import os
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

LIB = os.path.join(REPO, "oracle", "libhx_cpuref.so")


def _load():
    lib = ctypes.CDLL(LIB)
    fn = lib.hx_cpu_scan_agg
    fn.restype = ctypes.c_int
    fn.argtypes = [
        ctypes.POINTER(ctypes.c_char_p), ctypes.c_int,
        ctypes.c_int64, ctypes.c_int64, ctypes.c_int,
        ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_int64),
        ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int64),
        ctypes.POINTER(ctypes.c_double),
    ]
    return fn


def run_native(paths, ts_lo, ts_hi, threads=2):
    fn = _load()
    arr = (ctypes.c_char_p * len(paths))(*[p.encode() for p in paths])
    el = ctypes.c_double()
    rs = ctypes.c_int64()
    rm = ctypes.c_int64()
    ng = ctypes.c_int64()
    dg = ctypes.c_double()
    rc = fn(arr, len(paths), ts_lo, ts_hi, threads,
            ctypes.byref(el), ctypes.byref(rs), ctypes.byref(rm),
            ctypes.byref(ng), ctypes.byref(dg))
    assert rc == 0, f"native baseline rc={rc}"
    return dict(elapsed=el.value, rows_scanned=rs.value,
                rows_matched=rm.value, n_groups=ng.value, digest=dg.value)


@pytest.mark.skipif(not os.path.exists(LIB),
                    reason="oracle/libhx_cpuref.so not built")
@pytest.mark.parametrize("compression", ["none", "snappy"])
def test_native_baseline_matches_oracle(tmp_path, compression):
    import oracle
    from oracle.scan import AGG_SUM, AGG_COUNT
    from tools.gen_ssts import gen_dataset, middle_range

    d = str(tmp_path / compression)
    m = gen_dataset(d, n_rows=63_000, n_series=700, n_ssts=3, seed=11,
                    compression=compression)
    lo, hi = middle_range(m)
    ssts = [oracle.read_sst(s["path"]) for s in m["ssts"]]
    exp = oracle.scan_agg(ssts, (lo, hi), ops=AGG_SUM | AGG_COUNT)

    res = run_native([s["path"] for s in m["ssts"]], lo, hi)
    assert res["rows_matched"] == int(exp["count"].sum())
    assert res["n_groups"] == len(exp["series_id"])
    np.testing.assert_allclose(res["digest"], float(exp["sum"].sum()),
                               rtol=1e-9)
