# Append-mode GPU merge (VERDICT r01 missing #2 / SURVEY §8 operator
# surface): Binary value columns + BytesMergeOperator semantics
# (operator.rs:47-111, UpdateMode config.rs:166-172) through hx_scan —
# equal-PK rows' value bytes CONCATENATE in ascending __seq__ order.
# Pinned by the reference's own merge_stream golden (read.rs:512-573,
# tests/golden/merge_stream.json) mapped onto the metric PK shape, plus
# randomized parity vs the oracle's MERGE_APPEND restatement.
import json
import os
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

pytestmark = pytest.mark.gpu


def test_golden_merge_stream_append_through_hx_scan(tmp_path):
    # the reference's own golden vectors (read.rs:512-573): each input row
    # has its own seq => one single-row SST per row (file id = seq)
    from tools.gen_ssts import write_bytes_sst
    from horaedb_amd import Store
    with open(os.path.join(REPO, "tests", "golden",
                           "merge_stream.json")) as f:
        g = json.load(f)
    d = str(tmp_path / "store")
    for pk, val, seq in zip(g["input"]["pk1"], g["input"]["value"],
                            g["input"]["seq"]):
        write_bytes_sst(os.path.join(d, "data", f"{seq}.sst"),
                        [pk], [0], [val], seq)

    with Store(d, update_mode="append") as st:
        rows = st.scan((-10**15, 10**15), devices=[0])
    assert rows["series_id"].tolist() == g["expected_append"]["pk1"]
    assert [v.decode() for v in rows["value"]] == \
        g["expected_append"]["value"]

    # same store, Overwrite (LastValueOperator): last (highest-seq) wins
    with Store(d) as st:
        rows = st.scan((-10**15, 10**15), devices=[0])
    assert rows["series_id"].tolist() == g["expected_last"]["pk1"]
    assert [v.decode() for v in rows["value"]] == g["expected_last"]["value"]


def _rand_bytes_store(tmp_path, n_files=3, n_rows=3000, n_series=80):
    from tools.gen_ssts import write_bytes_sst
    d = str(tmp_path / "store")
    rng = np.random.default_rng(8)
    series_pool = np.sort(rng.integers(0, 2**63, n_series, dtype=np.uint64))
    for seq in range(1, n_files + 1):
        s = rng.choice(series_pool, size=n_rows)
        t = rng.integers(0, 40, n_rows) * 1000
        vals = [bytes(rng.integers(65, 90, rng.integers(1, 9)).astype(
            np.uint8).tolist()) for _ in range(n_rows)]
        write_bytes_sst(os.path.join(d, "data", f"{seq}.sst"), s, t, vals,
                        seq)
    return d


def _oracle_merge(d, ts_range, op):
    import glob
    import oracle
    from oracle.scan import merge_scan, MERGE_APPEND, MERGE_LAST
    ssts = [oracle.read_sst(p)
            for p in sorted(glob.glob(os.path.join(d, "data", "*.sst")),
                            key=lambda p: int(os.path.basename(p)[:-4]))]
    lo, hi = ts_range

    def pred(cols):
        return (cols[1] >= lo) & (cols[1] < hi)

    out = merge_scan(ssts, num_primary_keys=2,
                     merge_op=MERGE_APPEND if op == "append" else MERGE_LAST,
                     value_idxes=[2], predicate=pred)
    return out


def test_randomized_append_parity(tmp_path):
    from horaedb_amd import Store
    d = _rand_bytes_store(tmp_path)
    lo, hi = 5_000, 30_000
    with Store(d, update_mode="append") as st:
        rows = st.scan((lo, hi), devices=[0])
    exp = _oracle_merge(d, (lo, hi), "append")
    np.testing.assert_array_equal(rows["series_id"], exp[0])
    np.testing.assert_array_equal(rows["timestamp"], exp[1])
    assert rows["value"].tolist() == exp[2].tolist()


def test_randomized_overwrite_bytes_parity(tmp_path):
    # LastValueOperator over a Binary value column (Overwrite mode works
    # for any column type, operator.rs:37-44)
    from horaedb_amd import Store
    d = _rand_bytes_store(tmp_path)
    lo, hi = 0, 40_000
    with Store(d) as st:
        rows = st.scan((lo, hi), devices=[0])
    exp = _oracle_merge(d, (lo, hi), "last")
    np.testing.assert_array_equal(rows["series_id"], exp[0])
    np.testing.assert_array_equal(rows["timestamp"], exp[1])
    assert rows["value"].tolist() == exp[2].tolist()


def test_bytes_store_rejects_aggregate(tmp_path):
    from horaedb_amd import Store, HxError
    d = _rand_bytes_store(tmp_path, n_files=1, n_rows=100, n_series=10)
    with Store(d) as st:
        with pytest.raises(HxError) as ei:
            st.scan_agg((0, 10**9), devices=[0])
        assert ei.value.code == 3  # HX_ERR_UNSUPPORTED
