# CPU-side tests of the C-ABI library: it loads, exports every symbol
# include/horaedb_hx.h declares, the catalog (footer parse) matches pyarrow,
# and GPU-requiring calls fail loudly without a GPU (no CPU fallback).
import ctypes
import os
import re
import subprocess

import numpy as np
import pytest

import horaedb_amd
from horaedb_amd import Store, HxError

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _has_gpu():
    try:
        out = subprocess.run(["rocm-smi", "--showid"], capture_output=True,
                             timeout=10)
        return out.returncode == 0 and b"GPU" in out.stdout
    except Exception:
        return False


def test_library_exports_header_symbols():
    header = open(os.path.join(REPO, "include", "horaedb_hx.h")).read()
    decls = re.findall(r"^(?:hx_status|void|const char\*)\s+(hx_\w+)\(", header,
                       re.M)
    assert len(decls) >= 10, "header symbol scrape failed"
    lib = ctypes.CDLL(horaedb_amd.lib_path())
    for sym in decls:
        assert hasattr(lib, sym), f"symbol {sym} missing from libhoraedb_hx.so"


@pytest.fixture(scope="module")
def store_dir(tmp_path_factory):
    from tools.gen_ssts import gen_dataset
    out = str(tmp_path_factory.mktemp("capi"))
    m = gen_dataset(out, n_rows=50_000, n_series=250, n_ssts=4, seed=11)
    return out, m


def test_catalog_matches_pyarrow(store_dir):
    import pyarrow.parquet as pq
    out, m = store_dir
    with Store(out) as st:
        cat = st.catalog()
    assert len(cat) == len(m["ssts"])
    for entry, sst in zip(cat, sorted(m["ssts"], key=lambda s: s["seq"])):
        pf = pq.ParquetFile(sst["path"])
        assert entry["seq"] == sst["seq"]
        assert entry["n_rows"] == pf.metadata.num_rows
        assert entry["n_row_groups"] == pf.metadata.num_row_groups
        # ts min/max from our thrift stats parse == pyarrow statistics
        tmins, tmaxs = [], []
        for rg in range(pf.metadata.num_row_groups):
            c = pf.metadata.row_group(rg).column(1)
            tmins.append(c.statistics.min)
            tmaxs.append(c.statistics.max)
        assert entry["ts_min"] == min(tmins)
        assert entry["ts_max"] == max(tmaxs)


def test_find_ssts_time_overlap(store_dir):
    out, m = store_dir
    with Store(out) as st:
        # full range: all SSTs
        assert len(st.find_ssts((0, 2**62))) == len(m["ssts"])
        # empty range before data
        assert st.find_ssts((0, m["ts_start"])) == []
        # range covering exactly the first SST's window
        s0 = m["ssts"][0]
        hits = st.find_ssts((s0["ts_min"], s0["ts_max"] + 1))
        assert any(seq == s0["seq"] for _, seq in hits)
        # half-open end: range ending AT an sst's ts_min excludes it
        # (TimeRange semantics types.rs:125-127)
        s1 = m["ssts"][1]
        hits = st.find_ssts((m["ts_start"] - 10, s1["ts_min"]))
        assert all(seq != s1["seq"] for _, seq in hits)


@pytest.mark.skipif(_has_gpu(), reason="GPU present: NO_GPU path not testable")
def test_scan_fails_loudly_without_gpu(store_dir):
    out, m = store_dir
    with Store(out) as st:
        with pytest.raises(HxError) as ei:
            st.scan_agg((0, 2**62))
        assert ei.value.code == 4  # HX_ERR_NO_GPU
        assert "no CPU fallback" in str(ei.value)


def test_open_missing_store():
    with pytest.raises(HxError) as ei:
        Store("/nonexistent/path/xyz")
    assert ei.value.code == 1  # IO


def test_open_rejects_non_parquet(tmp_path):
    ddir = tmp_path / "data"
    ddir.mkdir()
    (ddir / "1.sst").write_bytes(b"this is not parquet at all........")
    with pytest.raises(HxError) as ei:
        Store(str(tmp_path))
    assert ei.value.code == 2  # FORMAT


def test_native_sst_writer_roundtrip(tmp_path):
    # the compaction output writer (csrc/parquet_writer.cpp): our footer +
    # pages must be readable by pyarrow AND by our own catalog reader
    import pyarrow.parquet as pq
    from horaedb_amd.store import write_sst_native

    n = 20_000
    rng = np.random.default_rng(3)
    series = np.sort(rng.integers(0, 500, n).astype(np.uint64))
    ts = np.arange(n, dtype=np.int64) * 7 + 1000
    val = rng.random(n)
    ddir = tmp_path / "data"
    ddir.mkdir()
    path = str(ddir / "9.sst")
    write_sst_native(path, series, ts, val, seq=9)

    t = pq.read_table(path)
    assert t.num_rows == n
    assert t.schema.names == ["series_id", "timestamp", "value", "__seq__",
                              "__reserved__"]
    np.testing.assert_array_equal(t.column("series_id").to_numpy(), series)
    np.testing.assert_array_equal(t.column("timestamp").to_numpy(), ts)
    np.testing.assert_array_equal(t.column("value").to_numpy(), val)
    assert t.column("__seq__").to_numpy().tolist() == [9] * n
    pf = pq.ParquetFile(path)
    assert pf.metadata.num_row_groups == (n + 8191) // 8192
    c = pf.metadata.row_group(0).column(1)
    assert c.statistics.min == 1000

    with Store(str(tmp_path)) as st:
        cat = st.catalog()
        assert len(cat) == 1
        assert cat[0]["seq"] == 9
        assert cat[0]["n_rows"] == n
        assert cat[0]["ts_min"] == 1000
        assert cat[0]["ts_max"] == int(ts.max())


def test_write_sort_and_catalog(tmp_path):
    # ColumnarStorage::write: unsorted batch -> PK-sorted SST + catalog add
    # (sort invariant golden, storage.rs:493-536 analog)
    import pyarrow.parquet as pq
    (tmp_path / "data").mkdir()
    with Store(str(tmp_path)) as st:
        seq = st.write([7, 5, 5, 9], [100, 200, 100, 50],
                       [1.0, 2.0, 3.0, 4.0])
        assert seq == 1
        cat = st.catalog()
        assert len(cat) == 1 and cat[0]["n_rows"] == 4
        t = pq.read_table(str(tmp_path / "data" / "1.sst"))
        assert t.column("series_id").to_numpy().tolist() == [5, 5, 7, 9]
        assert t.column("timestamp").to_numpy().tolist() == [100, 200, 100, 50]
        # second write gets the next file id
        seq2 = st.write([1], [10], [0.5])
        assert seq2 == 2
        assert len(st.find_ssts((0, 1000))) == 2


def test_write_stable_on_equal_pks(tmp_path):
    # equal PKs keep batch order (LastValueOperator last-wins contract)
    import pyarrow.parquet as pq
    (tmp_path / "data").mkdir()
    with Store(str(tmp_path)) as st:
        st.write([5, 5, 5], [100, 100, 100], [1.0, 2.0, 3.0])
    t = pq.read_table(str(tmp_path / "data" / "1.sst"))
    assert t.column("value").to_numpy().tolist() == [1.0, 2.0, 3.0]


def test_write_segment_crossing_check(tmp_path):
    (tmp_path / "data").mkdir()
    with Store(str(tmp_path), segment_duration_ms=1000) as st:
        with pytest.raises(HxError) as ei:
            st.write([1, 1], [100, 5000], [1.0, 2.0])
        assert ei.value.code == 6
        st.write([1, 1], [100, 5000], [1.0, 2.0], enable_check=False)


def test_schema_surface(store_dir):
    # ColumnarStorage::schema: the types.rs:150-240 contract
    out, m = store_dir
    with Store(out) as st:
        sch = st.schema()
    assert sch["num_primary_keys"] == 2
    assert [c["name"] for c in sch["columns"]] == \
        ["series_id", "timestamp", "value", "__seq__", "__reserved__"]
    assert [c["primary_key"] for c in sch["columns"]] == \
        [True, True, False, False, False]
    assert [c["builtin"] for c in sch["columns"]] == \
        [False, False, False, True, True]


def test_native_writer_odd_sizes(tmp_path):
    # row counts straddling row-group boundaries; single row; row_group 1
    import pyarrow.parquet as pq
    from horaedb_amd.store import write_sst_native
    rng = np.random.default_rng(7)
    for i, (n, rg) in enumerate([(1, 8192), (8191, 8192), (8192, 8192),
                                 (8193, 8192), (5, 1), (100, 7)]):
        series = np.sort(rng.integers(0, 50, n).astype(np.uint64))
        ts = np.arange(n, dtype=np.int64)
        val = rng.random(n)
        ddir = tmp_path / f"w{i}" / "data"
        ddir.mkdir(parents=True)
        p = str(ddir / "3.sst")
        write_sst_native(p, series, ts, val, seq=3, row_group=rg)
        t = pq.read_table(p)
        assert t.num_rows == n
        np.testing.assert_array_equal(t.column("series_id").to_numpy(),
                                      series)
        np.testing.assert_array_equal(t.column("value").to_numpy(), val)
        pf = pq.ParquetFile(p)
        assert pf.metadata.num_row_groups == (n + rg - 1) // rg
        with Store(str(tmp_path / f"w{i}")) as st:
            assert st.catalog()[0]["n_rows"] == n
