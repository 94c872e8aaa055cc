# The driver's bench entry starts with get_dataset: guard generation +
# reuse + oracle readability on a tiny config (CPU-only).
import json
import os
import sys
import types

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def _args(tmp, rows=20_000, series=100, ssts=4):
    a = types.SimpleNamespace()
    a.rows, a.series, a.ssts = rows, series, ssts
    a.seed = 7
    a.compression = "none"
    a.ts_encoding = "PLAIN"
    a.data_dir = str(tmp)
    return a


def test_get_dataset_generate_reuse_and_scan(tmp_path):
    import bench
    import oracle
    from oracle.scan import AGG_SUM, AGG_COUNT
    from tools.gen_ssts import middle_range

    out, m = bench.get_dataset(_args(tmp_path), rank=0)
    assert m["n_rows"] == 20_000 and m["n_ssts"] == 4
    # reuse hits the cache (same tag)
    out2, m2 = bench.get_dataset(_args(tmp_path), rank=0)
    assert out2 == out and m2["n_rows"] == m["n_rows"]
    # a different rank gets a different seed => different dir
    out3, _ = bench.get_dataset(_args(tmp_path), rank=1)
    assert out3 != out

    # the generated store is oracle-readable and internally consistent
    ddir = os.path.join(out, "data")
    ssts = [oracle.read_sst(os.path.join(ddir, f))
            for f in sorted(os.listdir(ddir)) if f.endswith(".sst")]
    assert sum(s.n_rows for s in ssts) == 20_000
    res = oracle.scan_agg(ssts, middle_range(m), ops=AGG_SUM | AGG_COUNT)
    assert 0 < len(res["series_id"]) <= 100
    assert int(res["count"].sum()) > 0
    # SSTs are PK-sorted (writer invariant; series stays u64 — ids span
    # the full u64 range)
    for s in ssts:
        sid, ts = s.cols[0], s.cols[1]
        adj = (sid[:-1] < sid[1:]) | ((sid[:-1] == sid[1:]) &
                                      (ts[:-1] <= ts[1:]))
        assert bool(np.all(adj))
