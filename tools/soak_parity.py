#!/usr/bin/env python3
# Randomized parity soak (GPU): random shapes x codecs x encodings x
# predicates x ops, engine vs oracle, until the time budget runs out.
# Exercises: snappy/zstd/uncompressed, PLAIN/DELTA ts, overlap generations,
# series sets, buckets, append/byte stores, partial compaction.
import os
import random
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import numpy as np  # noqa: E402


def one_case(rng, tmp, i):
    import oracle
    from oracle.scan import AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX
    from tools.gen_ssts import gen_dataset, middle_range
    from horaedb_amd import Store

    n_series = int(rng.choice([37, 500, 4096, 50_000]))
    pts = int(rng.choice([2, 8, 30]))
    n_rows = n_series * pts
    n_ssts = int(rng.choice([1, 3, 7, 16]))
    comp = str(rng.choice(["none", "snappy", "zstd"]))
    tse = str(rng.choice(["PLAIN", "DELTA_BINARY_PACKED"]))
    gens = int(rng.choice([1, 1, 2]))
    d = os.path.join(tmp, f"case{i}")
    m = gen_dataset(d, n_rows, n_series, n_ssts, seed=1000 + i,
                    compression=comp, ts_encoding=tse, overlap_gens=gens)
    frac = float(rng.choice([0.25, 0.5, 1.0]))
    lo, hi = middle_range(m, frac)
    ops = AGG_SUM | AGG_COUNT | AGG_MIN | AGG_MAX
    series_in = None
    if rng.random() < 0.4:
        ids = np.load(os.path.join(d, "series_ids.npy"))
        k = max(1, int(len(ids) * rng.choice([0.01, 0.2])))
        series_in = rng.choice(ids, size=k, replace=False).tolist()
    bucket_ms = int(rng.choice([0, 0, 60_000]))

    with Store(d) as st:
        res = st.scan_agg((lo, hi), ops=ops, bucket_ms=bucket_ms,
                          series_in=series_in, devices=[0])
    ssts = [oracle.read_sst(s["path"]) for s in m["ssts"]]
    sset = set(int(x) for x in series_in) if series_in else None
    exp = oracle.scan_agg(ssts, (lo, hi), ops=ops, bucket_ms=bucket_ms,
                          series_set=sset)
    assert res["series_id"].tolist() == exp["series_id"].tolist(), \
        f"case {i}: keys differ ({comp},{tse},gens={gens})"
    if bucket_ms:
        assert res["bucket"].tolist() == exp["bucket"].tolist()
    np.testing.assert_array_equal(res["count"], exp["count"])
    np.testing.assert_array_equal(res["vmin"], exp["vmin"])
    np.testing.assert_array_equal(res["vmax"], exp["vmax"])
    np.testing.assert_allclose(res["sum"], exp["sum"], rtol=1e-9)
    return f"{n_rows}r/{n_ssts}f/{comp}/{tse}/g{gens}/b{bucket_ms}" + \
        (f"/sel{len(series_in)}" if series_in else "")


def main():
    budget = float(sys.argv[1]) if len(sys.argv) > 1 else 300
    import tempfile
    t0 = time.time()
    n = 0
    rng = np.random.default_rng(int(sys.argv[2]) if len(sys.argv) > 2 else 7)
    with tempfile.TemporaryDirectory() as tmp:
        while time.time() - t0 < budget:
            desc = one_case(rng, tmp, n)
            n += 1
            print(f"[soak] case {n} OK: {desc}", flush=True)
            if n % 8 == 0:
                import shutil
                for e in os.listdir(tmp):
                    shutil.rmtree(os.path.join(tmp, e), ignore_errors=True)
    print(f"[soak] PASSED {n} randomized parity cases in "
          f"{time.time() - t0:.0f}s")


if __name__ == "__main__":
    main()
