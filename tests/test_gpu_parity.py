# GPU parity tests: the product path (libhoraedb_hx.so HIP kernels) against
# the oracle (numpy/pyarrow restatement) on the same seeded SSTs.
# Bar (BASELINE.json): counts/min/max bit-exact; f64 sums <= 1e-9 relative.
import os

import numpy as np
import pytest

import oracle
from oracle.scan import AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, AGG_AVG
from tools.gen_ssts import gen_dataset, gen_sst_from_arrays, middle_range

pytestmark = pytest.mark.gpu

OPS_ALL = AGG_SUM | AGG_COUNT | AGG_MIN | AGG_MAX | AGG_AVG


def _store():
    from horaedb_amd import Store
    return Store


def check_parity(store_dir, ts_range, ops=OPS_ALL, bucket_ms=0,
                 series_in=None, sst_paths=None):
    from horaedb_amd import Store
    with Store(store_dir) as st:
        res = st.scan_agg(ts_range, ops=ops, bucket_ms=bucket_ms,
                          series_in=series_in, devices=[0])
    if sst_paths is None:
        ddir = os.path.join(store_dir, "data")
        sst_paths = sorted(
            (os.path.join(ddir, f) for f in os.listdir(ddir)
             if f.endswith(".sst")),
            key=lambda p: int(os.path.basename(p).split(".")[0]))
    ssts = [oracle.read_sst(p) for p in sst_paths]
    exp = oracle.scan_agg(ssts, ts_range, series_set=series_in,
                          bucket_ms=bucket_ms, ops=ops)
    assert res["series_id"].tolist() == exp["series_id"].tolist(), \
        "group keys differ"
    if bucket_ms:
        assert res["bucket"].tolist() == exp["bucket"].tolist()
    if ops & AGG_COUNT:
        assert res["count"].tolist() == exp["count"].tolist()
    if ops & AGG_MIN:
        np.testing.assert_array_equal(res["vmin"], exp["vmin"])
    if ops & AGG_MAX:
        np.testing.assert_array_equal(res["vmax"], exp["vmax"])
    if ops & AGG_SUM:
        np.testing.assert_allclose(res["sum"], exp["sum"], rtol=1e-9)
    if ops & AGG_AVG:
        np.testing.assert_allclose(res["avg"], exp["avg"], rtol=1e-9)
    return res


@pytest.fixture(scope="module")
def ds_plain(tmp_path_factory):
    out = str(tmp_path_factory.mktemp("plain"))
    m = gen_dataset(out, n_rows=400_000, n_series=2_000, n_ssts=8, seed=42)
    return out, m


@pytest.fixture(scope="module")
def ds_snappy(tmp_path_factory):
    out = str(tmp_path_factory.mktemp("snappy"))
    m = gen_dataset(out, n_rows=200_000, n_series=1_000, n_ssts=4, seed=44,
                    compression="snappy")
    return out, m


@pytest.fixture(scope="module")
def ds_delta(tmp_path_factory):
    out = str(tmp_path_factory.mktemp("delta"))
    m = gen_dataset(out, n_rows=200_000, n_series=1_000, n_ssts=4, seed=43,
                    ts_encoding="DELTA_BINARY_PACKED")
    return out, m


def test_full_range_sum_count(ds_plain):
    out, m = ds_plain
    check_parity(out, (0, 2**62), ops=AGG_SUM | AGG_COUNT)


def test_middle_range_all_ops(ds_plain):
    out, m = ds_plain
    check_parity(out, middle_range(m), ops=OPS_ALL)


def test_narrow_range_crossing_row_groups(ds_plain):
    out, m = ds_plain
    lo = m["ts_start"] + 37 * m["step_ms"]
    hi = m["ts_start"] + 38 * m["step_ms"]  # single point per series
    check_parity(out, (lo, hi), ops=AGG_SUM | AGG_COUNT)


def test_empty_range(ds_plain):
    out, m = ds_plain
    res = check_parity(out, (17, 18))
    assert len(res["series_id"]) == 0


def test_series_set_predicate(ds_plain):
    out, m = ds_plain
    ids = np.load(os.path.join(out, "series_ids.npy"))
    sel = ids[:: 50].tolist()  # 2% of series
    check_parity(out, middle_range(m), ops=AGG_MIN | AGG_MAX | AGG_AVG,
                 series_in=sel)


def test_time_bucket(ds_plain):
    out, m = ds_plain
    check_parity(out, middle_range(m), ops=AGG_SUM | AGG_COUNT,
                 bucket_ms=60_000)


def test_delta_encoded_timestamps(ds_delta):
    out, m = ds_delta
    check_parity(out, middle_range(m), ops=OPS_ALL)
    check_parity(out, (0, 2**62), ops=AGG_SUM | AGG_COUNT)


def test_snappy_pages(ds_snappy):
    # the reference's DEFAULT codec (config.rs:120-133): GPU decompress
    out, m = ds_snappy
    check_parity(out, middle_range(m), ops=OPS_ALL)
    check_parity(out, (0, 2**62), ops=AGG_SUM | AGG_COUNT)


def test_rle_dictionary(tmp_path):
    # RLE_DICTIONARY chunks (dict page + RLE/bit-packed hybrid indices)
    store = str(tmp_path)
    rng = np.random.default_rng(7)
    n = 60_000
    series = np.sort(rng.integers(0, 300, n).astype(np.uint64))
    ts = np.arange(n, dtype=np.int64) * 10
    vals = rng.random(n)
    gen_sst_from_arrays(store, 1, series, ts, vals,
                        dict_columns=("series_id", "value"))
    check_parity(store, (0, 10**9), ops=OPS_ALL)
    check_parity(store, (100_000, 400_000), ops=AGG_SUM | AGG_COUNT)


def test_rle_dictionary_snappy(tmp_path):
    store = str(tmp_path)
    rng = np.random.default_rng(8)
    n = 30_000
    series = np.sort(rng.integers(0, 100, n).astype(np.uint64))
    ts = np.arange(n, dtype=np.int64) * 10
    vals = rng.random(n)
    gen_sst_from_arrays(store, 1, series, ts, vals, compression="snappy",
                        dict_columns=("series_id",))
    check_parity(store, (0, 10**9), ops=AGG_SUM | AGG_COUNT)


def test_snappy_delta_combined(tmp_path_factory):
    out = str(tmp_path_factory.mktemp("snapdelta"))
    m = gen_dataset(out, n_rows=100_000, n_series=500, n_ssts=2, seed=45,
                    compression="snappy", ts_encoding="DELTA_BINARY_PACKED")
    check_parity(out, middle_range(m), ops=AGG_SUM | AGG_COUNT)


# ---------------------------------------------------------------------------
# dedup scenarios (MergeExec parity)
# ---------------------------------------------------------------------------

def test_dedup_within_sst(tmp_path):
    store = str(tmp_path)
    # duplicate PKs inside one SST: last row wins (LastValueOperator)
    gen_sst_from_arrays(store, 1, [5, 5, 5, 7], [100, 100, 200, 100],
                        [1.0, 2.0, 3.0, 4.0], sort=False)
    res = check_parity(store, (0, 1000), ops=AGG_SUM | AGG_COUNT)
    assert res["count"].tolist() == [2, 1]
    np.testing.assert_allclose(res["sum"], [2.0 + 3.0, 4.0])


def test_dedup_across_ssts_newer_wins(tmp_path):
    store = str(tmp_path)
    gen_sst_from_arrays(store, 1, [5, 5, 7], [100, 200, 100], [1.0, 2.0, 3.0])
    gen_sst_from_arrays(store, 2, [5, 7, 8], [200, 100, 50],
                        [20.0, 30.0, 40.0])
    res = check_parity(store, (0, 1000), ops=AGG_SUM | AGG_COUNT)
    np.testing.assert_allclose(res["sum"], [1.0 + 20.0, 30.0, 40.0])


def test_dedup_three_overlapping_ssts(tmp_path):
    store = str(tmp_path)
    rng = np.random.default_rng(9)
    for seq in (1, 2, 3):
        n = 5000
        series = rng.integers(0, 50, n).astype(np.uint64)
        ts = rng.integers(0, 200, n).astype(np.int64) * 10
        vals = rng.random(n)
        gen_sst_from_arrays(store, seq, series, ts, vals)
    check_parity(store, (0, 10**6), ops=OPS_ALL)
    check_parity(store, (500, 1500), ops=AGG_SUM | AGG_COUNT)


def test_dedup_duplicate_on_row_group_boundary(tmp_path):
    store = str(tmp_path)
    # rows 8191 and 8192 share a PK -> dedup across the row-group boundary
    n = 8192 + 64
    series = (np.arange(n, dtype=np.uint64) + 1) // 2
    ts = np.full(n, 500, dtype=np.int64)
    vals = np.arange(n, dtype=np.float64)
    gen_sst_from_arrays(store, 1, series, ts, vals, sort=False)
    res = check_parity(store, (0, 1000), ops=AGG_SUM | AGG_COUNT)
    assert series[8191] == series[8192]  # the boundary pair the test is about
    assert res["count"].tolist() == [1] * len(res["series_id"])


def test_disjoint_ts_ssts_no_dedup(tmp_path):
    store = str(tmp_path)
    # same series, disjoint ts windows: no dedup (union of segments)
    gen_sst_from_arrays(store, 1, [5, 7], [100, 100], [1.0, 2.0])
    gen_sst_from_arrays(store, 2, [5, 7], [900, 900], [10.0, 20.0])
    res = check_parity(store, (0, 1000), ops=AGG_SUM | AGG_COUNT)
    assert res["count"].tolist() == [2, 2]


def test_multi_device_if_available(ds_plain):
    out, m = ds_plain
    import torch
    n_gpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
    if n_gpu < 2:
        pytest.skip("single GPU box")
    from horaedb_amd import Store
    with Store(out) as st:
        res = st.scan_agg(middle_range(m), devices=[0, 1])
    ddir = os.path.join(out, "data")
    paths = sorted((os.path.join(ddir, f) for f in os.listdir(ddir)
                    if f.endswith(".sst")),
                   key=lambda p: int(os.path.basename(p).split(".")[0]))
    ssts = [oracle.read_sst(p) for p in paths]
    exp = oracle.scan_agg(ssts, middle_range(m))
    assert res["series_id"].tolist() == exp["series_id"].tolist()
    np.testing.assert_allclose(res["sum"], exp["sum"], rtol=1e-9)


# ---------------------------------------------------------------------------
# streaming parity mode (hx_scan): row-level merged output
# ---------------------------------------------------------------------------

def check_scan_rows(store_dir, ts_range, series_in=None, segment_ms=0):
    from horaedb_amd import Store
    kw = {"segment_duration_ms": segment_ms} if segment_ms else {}
    with Store(store_dir, **kw) as st:
        res = st.scan(ts_range, series_in=series_in, devices=[0])
    ddir = os.path.join(store_dir, "data")
    paths = sorted((os.path.join(ddir, f) for f in os.listdir(ddir)
                    if f.endswith(".sst")),
                   key=lambda p: int(os.path.basename(p).split(".")[0]))
    ssts = [oracle.read_sst(p) for p in paths]
    exp = oracle.scan_rows(ssts, ts_range, series_set=series_in,
                           segment_ms=segment_ms or 12 * 3600 * 1000)
    assert res["series_id"].tolist() == exp["series_id"].tolist()
    assert res["timestamp"].tolist() == exp["timestamp"].tolist()
    np.testing.assert_array_equal(res["value"], exp["value"])
    return res


def test_scan_rows_basic(ds_plain):
    out, m = ds_plain
    check_scan_rows(out, middle_range(m))


def test_scan_rows_dedup(tmp_path):
    store = str(tmp_path)
    gen_sst_from_arrays(store, 1, [5, 5, 7], [100, 200, 100], [1.0, 2.0, 3.0])
    gen_sst_from_arrays(store, 2, [5, 7], [200, 100], [20.0, 30.0])
    res = check_scan_rows(store, (0, 1000))
    assert res["value"].tolist() == [1.0, 20.0, 30.0]


def test_scan_rows_segments(tmp_path):
    # two time segments: output is PER-SEGMENT sorted (segment-major), not
    # globally PK-sorted (the reference's per-segment union order)
    store = str(tmp_path)
    gen_sst_from_arrays(store, 1, [9, 5], [100, 150], [1.0, 2.0])
    gen_sst_from_arrays(store, 2, [5, 9], [1100, 1150], [3.0, 4.0])
    res = check_scan_rows(store, (0, 5000), segment_ms=1000)
    assert res["series_id"].tolist() == [5, 9, 5, 9]
    assert res["value"].tolist() == [2.0, 1.0, 3.0, 4.0]


def test_scan_rows_projection(ds_plain):
    from horaedb_amd import Store
    out, m = ds_plain
    with Store(out) as st:
        res = st.scan(middle_range(m), projection=[2, 0], devices=[0])
    assert set(res.keys()) == {"value", "series_id"}


# ---------------------------------------------------------------------------
# compaction: GPU merge-dedup -> one new SST
# ---------------------------------------------------------------------------

def test_compact_overlapping_ssts(tmp_path):
    from horaedb_amd import Store
    store = str(tmp_path)
    gen_sst_from_arrays(store, 1, [5, 5, 7], [100, 200, 100], [1.0, 2.0, 3.0])
    gen_sst_from_arrays(store, 2, [5, 7, 8], [200, 100, 50],
                        [20.0, 30.0, 40.0])
    # oracle BEFORE compaction (the inputs define the expected result)
    ddir = os.path.join(store, "data")
    ssts = [oracle.read_sst(os.path.join(ddir, f"{q}.sst")) for q in (1, 2)]
    exp = oracle.scan_agg(ssts, (0, 10**6), ops=AGG_SUM | AGG_COUNT)
    exp_rows = oracle.scan_rows(ssts, (0, 10**6))

    with Store(store) as st:
        new_seq = st.compact((0, 10**6), devices=[0])
        assert new_seq == 3
        cat = st.catalog()
        assert [c["seq"] for c in cat] == [3]
        assert cat[0]["n_rows"] == 4  # 5 rows deduped to 4
        # post-compaction scans reproduce the pre-compaction results
        res = st.scan_agg((0, 10**6), ops=AGG_SUM | AGG_COUNT, devices=[0])
        assert res["series_id"].tolist() == exp["series_id"].tolist()
        np.testing.assert_allclose(res["sum"], exp["sum"], rtol=1e-9)
        assert res["count"].tolist() == exp["count"].tolist()
        rows = st.scan((0, 10**6), devices=[0])
        assert rows["series_id"].tolist() == exp_rows["series_id"].tolist()
        np.testing.assert_array_equal(rows["value"], exp_rows["value"])
    assert sorted(os.listdir(ddir)) == ["3.sst"]
    # the output is standard parquet (readable by pyarrow)
    import pyarrow.parquet as pq
    t = pq.read_table(os.path.join(ddir, "3.sst"))
    assert t.num_rows == 4


def test_compact_larger_randomized(tmp_path):
    from horaedb_amd import Store
    store = str(tmp_path)
    rng = np.random.default_rng(11)
    for seq in (1, 2, 3):
        n = 30_000
        series = rng.integers(0, 400, n).astype(np.uint64)
        ts = rng.integers(0, 5000, n).astype(np.int64) * 10
        gen_sst_from_arrays(store, seq, series, ts, rng.random(n))
    ddir = os.path.join(store, "data")
    ssts = [oracle.read_sst(os.path.join(ddir, f"{q}.sst"))
            for q in (1, 2, 3)]
    exp = oracle.scan_agg(ssts, (0, 10**9), ops=OPS_ALL)
    with Store(store) as st:
        new_seq = st.compact((0, 10**9), devices=[0])
        assert new_seq == 4
        res = st.scan_agg((0, 10**9), ops=OPS_ALL, devices=[0])
    assert res["series_id"].tolist() == exp["series_id"].tolist()
    assert res["count"].tolist() == exp["count"].tolist()
    np.testing.assert_array_equal(res["vmin"], exp["vmin"])
    np.testing.assert_array_equal(res["vmax"], exp["vmax"])
    np.testing.assert_allclose(res["sum"], exp["sum"], rtol=1e-9)


def test_compact_leaves_disjoint_files(tmp_path):
    from horaedb_amd import Store
    store = str(tmp_path)
    gen_sst_from_arrays(store, 1, [5], [100], [1.0])
    gen_sst_from_arrays(store, 2, [5], [150], [2.0])
    gen_sst_from_arrays(store, 3, [5], [10**7], [3.0])  # far away in time
    with Store(store) as st:
        new_seq = st.compact((0, 1000), devices=[0])
        assert new_seq == 4
        cat = st.catalog()
        assert sorted(c["seq"] for c in cat) == [3, 4]
        res = st.scan_agg((0, 2 * 10**7), ops=AGG_SUM | AGG_COUNT, devices=[0])
        assert res["count"].tolist() == [3]
        np.testing.assert_allclose(res["sum"], [6.0])


def test_write_then_scan_reference_golden(tmp_path):
    # the reference's own end-to-end golden (storage.rs:392-491) mapped onto
    # the metric schema (pk1 -> series_id, pk2 -> timestamp): two writes into
    # one segment, Overwrite dedup, predicate applied before the merge
    from horaedb_amd import Store
    (tmp_path / "data").mkdir()
    with Store(str(tmp_path)) as st:
        st.write([11, 11, 9, 10, 5], [100, 100, 1, 2, 3], [2.0, 7.0, 4.0, 6.0, 1.0])
        st.write([11, 11, 9, 10], [100, 99, 1, 2], [22.0, 77.0, 44.0, 66.0])
        rows = st.scan((0, 2**40), devices=[0])
        # expected (storage.rs:448-461): (5,3,1)(9,1,44)(10,2,66)(11,99,77)(11,100,22)
        assert rows["series_id"].tolist() == [5, 9, 10, 11, 11]
        assert rows["timestamp"].tolist() == [3, 1, 2, 99, 100]
        assert rows["value"].tolist() == [1.0, 44.0, 66.0, 77.0, 22.0]
        # predicate pk1 == 11 (storage.rs:475-489)
        rows = st.scan((0, 2**40), series_in=[11], devices=[0])
        assert rows["series_id"].tolist() == [11, 11]
        assert rows["value"].tolist() == [77.0, 22.0]


# ---------------------------------------------------------------------------
# randomized property parity: arbitrary small stores
# ---------------------------------------------------------------------------

def test_randomized_stores_parity(tmp_path_factory):
    rng = np.random.default_rng(12345)
    for trial in range(6):
        store = str(tmp_path_factory.mktemp(f"rand{trial}"))
        n_ssts = int(rng.integers(1, 4))
        for seq in range(1, n_ssts + 1):
            n = int(rng.integers(1, 20_000))
            series = rng.integers(0, int(rng.integers(1, 1000)), n).astype(np.uint64)
            ts = rng.integers(0, int(rng.integers(10, 100_000)), n).astype(np.int64)
            gen_sst_from_arrays(store, seq, series, ts, rng.random(n))
        lo = int(rng.integers(0, 50_000))
        hi = lo + int(rng.integers(1, 100_000))
        check_parity(store, (lo, hi), ops=OPS_ALL)
        check_parity(store, (lo, hi), ops=AGG_SUM | AGG_COUNT,
                     bucket_ms=int(rng.integers(1, 5000)))


def test_delta_wide_bitwidths(tmp_path):
    # delta pages with huge jumps (wide miniblock bit widths incl. >32)
    store = str(tmp_path)
    rng = np.random.default_rng(77)
    n = 30_000
    series = np.sort(rng.integers(0, 100, n).astype(np.uint64))
    ts = np.cumsum(rng.integers(-2**40, 2**40, n)).astype(np.int64)
    order = np.lexsort((ts, series))
    gen_sst_from_arrays(store, 1, series[order], ts[order], rng.random(n),
                        sort=False, ts_encoding="DELTA_BINARY_PACKED")
    lo, hi = int(ts.min()), int(ts.max()) + 1
    check_parity(store, (lo, hi), ops=AGG_SUM | AGG_COUNT)


def test_negative_timestamps_buckets_and_segments(tmp_path):
    # floor-division semantics for negative ts (truncate_by types.rs:82-86)
    store = str(tmp_path)
    rng = np.random.default_rng(21)
    n = 20_000
    series = np.sort(rng.integers(0, 200, n).astype(np.uint64))
    ts = rng.integers(-100_000, 100_000, n).astype(np.int64)
    order = np.lexsort((ts, series))
    gen_sst_from_arrays(store, 1, series[order], ts[order], rng.random(n),
                        sort=False)
    check_parity(store, (-200_000, 200_000), ops=AGG_SUM | AGG_COUNT,
                 bucket_ms=7_000)
    check_parity(store, (-50_000, 50_000), ops=OPS_ALL)
    # streaming mode with negative segments
    check_scan_rows(store, (-200_000, 200_000), segment_ms=30_000)

def test_data_page_v2(tmp_path):
    # DataPageHeaderV2 (REQUIRED flat columns => zero level bytes); the
    # walker's v2 branch + payload offset math
    from tools.gen_ssts import write_sst
    store = str(tmp_path)
    os.makedirs(os.path.join(store, "data"))
    rng = np.random.default_rng(21)
    n = 50_000
    series = np.sort(rng.integers(0, 400, n).astype(np.uint64))
    ts = np.arange(n, dtype=np.int64) * 9
    vals = rng.random(n)
    write_sst(os.path.join(store, "data", "1.sst"), series, ts, vals, 1,
              data_page_version="2.0")
    check_parity(store, (0, 10**9), ops=OPS_ALL)
    check_parity(store, (50_000, 300_000), ops=AGG_SUM | AGG_COUNT)


def test_data_page_v2_snappy(tmp_path):
    # v2 + snappy: levels (0 bytes here) stay uncompressed, data compressed
    from tools.gen_ssts import write_sst
    store = str(tmp_path)
    os.makedirs(os.path.join(store, "data"))
    rng = np.random.default_rng(22)
    n = 40_000
    series = np.sort(rng.integers(0, 200, n).astype(np.uint64))
    ts = np.arange(n, dtype=np.int64) * 5
    vals = rng.random(n)
    write_sst(os.path.join(store, "data", "1.sst"), series, ts, vals, 1,
              compression="snappy", data_page_version="2.0")
    check_parity(store, (0, 10**9), ops=AGG_SUM | AGG_COUNT | AGG_MIN)


def test_unsupported_codec_fails_loudly(tmp_path):
    # out-of-scope codec => HX_ERR_UNSUPPORTED, never silent wrong numbers
    from horaedb_amd import Store, HxError
    from tools.gen_ssts import write_sst
    store = str(tmp_path)
    os.makedirs(os.path.join(store, "data"))
    n = 10_000
    write_sst(os.path.join(store, "data", "1.sst"),
              np.arange(n, dtype=np.uint64), np.arange(n, dtype=np.int64),
              np.ones(n), 1, compression="gzip")
    with Store(store) as st:
        with pytest.raises(HxError) as ei:
            st.scan_agg((0, 10**9), devices=[0])
        assert ei.value.code == 3  # HX_ERR_UNSUPPORTED
        assert "codec" in str(ei.value)


def test_zstd_pages_parity(tmp_path):
    # Zstd (config.rs:84): pages are decompressed on the HOST at staging
    # (DESIGN §2 — the codec is inherently serial; staging is untimed),
    # kernels then scan the raw pages. Full parity vs the oracle.
    from tools.gen_ssts import gen_dataset, middle_range
    out = str(tmp_path / "z")
    m = gen_dataset(out, n_rows=60_000, n_series=600, n_ssts=3, seed=33,
                    compression="zstd")
    check_parity(out, middle_range(m),
                 ops=AGG_SUM | AGG_COUNT | AGG_MIN | AGG_MAX)


def test_series_set_unsorted_with_absent_ids(ds_plain):
    # predicate ids arrive unsorted, with duplicates and ids not in the data
    out, m = ds_plain
    ids = np.load(os.path.join(out, "series_ids.npy"))
    sel = ids[::71].tolist()
    sel = sel[::-1] + sel[:3] + [2**63 + 5, 12345]  # reversed + dups + absent
    check_parity(out, middle_range(m), ops=AGG_SUM | AGG_COUNT,
                 series_in=sel)


def test_many_tiny_overlapping_ssts(tmp_path):
    # 40 single-row-group SSTs over the same ts window: dedup across the
    # whole cluster, newest file wins per PK
    store = str(tmp_path)
    rng = np.random.default_rng(23)
    for seq in range(1, 41):
        n = 50
        series = rng.integers(0, 25, n).astype(np.uint64)
        ts = rng.integers(0, 2_000, n).astype(np.int64)
        vals = rng.random(n)
        gen_sst_from_arrays(store, seq, series, ts, vals)
    check_parity(store, (0, 10**9), ops=OPS_ALL)
    check_parity(store, (500, 1_500), ops=AGG_SUM | AGG_COUNT, bucket_ms=250)


def test_scan_early_stop_batch_limit(ds_plain):
    # hx_scan's callback contract: nonzero return stops the stream cleanly
    # (horaedb_hx.h hx_batch_cb; the reference consumer dropping the stream)
    from horaedb_amd import Store
    out, m = ds_plain
    with Store(out) as st:
        full = st.scan((0, 2**62), devices=[0])
        part = st.scan((0, 2**62), devices=[0], batch_limit=2)
    assert len(full["series_id"]) > 2 * 65536
    assert len(part["series_id"]) == 2 * 65536
    np.testing.assert_array_equal(part["series_id"],
                                  full["series_id"][:2 * 65536])
    np.testing.assert_array_equal(part["value"], full["value"][:2 * 65536])


def test_concurrent_prepared_scans(ds_plain):
    # the concurrency contract (horaedb_hx.h): scan-side calls from many
    # threads of one handle, one call per prepared — two prepareds on two
    # threads must produce identical, independent results (the server
    # concurrent-scan pattern bench.py --pipeline relies on)
    from concurrent.futures import ThreadPoolExecutor
    from horaedb_amd import Store
    out, m = ds_plain
    with Store(out) as st:
        preps = [st.prepare(middle_range(m), devices=[0]) for _ in range(3)]
        try:
            def run(p):
                return p.exec_agg(ops=OPS_ALL)
            for _ in range(3):  # a few rounds to shake out races
                with ThreadPoolExecutor(3) as ex:
                    rs = list(ex.map(run, preps))
                for r in rs[1:]:
                    assert r["series_id"].tolist() == rs[0]["series_id"].tolist()
                    assert r["count"].tolist() == rs[0]["count"].tolist()
                    np.testing.assert_array_equal(r["vmin"], rs[0]["vmin"])
                    np.testing.assert_allclose(r["sum"], rs[0]["sum"],
                                               rtol=1e-12)
        finally:
            for p in preps:
                p.close()


def test_wave_kernel_fallback_parity(ds_plain):
    # HX_RANGE=0 must route through k_scan_agg (the wave kernel) with
    # identical results — the fallback stays load-bearing for bucket/state
    # paths and must not rot
    out, m = ds_plain
    os.environ["HX_RANGE"] = "0"
    try:
        check_parity(out, middle_range(m), ops=OPS_ALL)
        check_parity(out, (0, 2**62), ops=AGG_SUM | AGG_COUNT)
    finally:
        del os.environ["HX_RANGE"]


def test_invalid_agg_args(ds_plain):
    # argument validation surfaces HX_ERR_INVALID, not kernel misbehavior
    from horaedb_amd import Store, HxError
    out, m = ds_plain
    with Store(out) as st:
        p = st.prepare(middle_range(m), devices=[0])
        try:
            with pytest.raises(HxError) as ei:
                p.exec_agg(ops=0)
            assert ei.value.code == 6
            with pytest.raises(HxError) as ei:
                p.exec_agg(ops=AGG_SUM, bucket_ms=-5)
            assert ei.value.code == 6
            # still usable after rejected calls
            r = p.exec_agg(ops=AGG_SUM | AGG_COUNT)
            assert len(r["series_id"]) > 0
        finally:
            p.close()


def test_corrupt_page_header_fails_loudly(tmp_path):
    # a corrupted page header inside a chunk must fail staging with a real
    # error (FORMAT/UNSUPPORTED), never feed garbage to the kernels
    from horaedb_amd import Store, HxError
    from tools.gen_ssts import write_sst
    store = str(tmp_path)
    os.makedirs(os.path.join(store, "data"))
    p = os.path.join(store, "data", "1.sst")
    n = 30_000
    write_sst(p, np.arange(n, dtype=np.uint64), np.arange(n, dtype=np.int64),
              np.ones(n), 1)
    blob = bytearray(open(p, "rb").read())
    # clobber bytes shortly after the leading magic: inside the first
    # page header / payload region
    for off in (8, 9, 10, 11):
        blob[off] ^= 0xFF
    open(p, "wb").write(bytes(blob))
    with Store(store) as st:
        try:
            r = st.scan_agg((0, 2**62), ops=AGG_SUM | AGG_COUNT,
                            devices=[0])
            # parser may legitimately treat flipped VALUE bytes as data;
            # but if it returns, the result must still be well-formed
            assert len(r["series_id"]) <= n
        except HxError as e:
            assert e.code in (2, 3, 5, 7)
