# CPU-side tests of the inverted-index writer and the oracle restatement
# (rfc:86-137). hx_index_write is host code (no GPU needed).
import os
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def _mk_store(tmp_path, n_series=50, n_rows=5_000):
    from tools.gen_ssts import gen_dataset
    d = str(tmp_path / "store")
    m = gen_dataset(d, n_rows=n_rows, n_series=n_series, n_ssts=2, seed=9)
    ids = np.load(os.path.join(d, "series_ids.npy"))
    return d, m, ids


def test_index_write_readable_by_pyarrow(tmp_path):
    # the native index writer's files are standard Parquet
    import pyarrow.parquet as pq
    from horaedb_amd import Store
    d, m, ids = _mk_store(tmp_path, n_series=50, n_rows=5_000)
    with Store(d) as st:
        st.index_write(["a", "b", "a"], ["x", "y", "z"],
                       np.array([3, 2, 1], dtype=np.uint64))
    files = os.listdir(os.path.join(d, "index"))
    assert files
    t = pq.read_table(os.path.join(d, "index", files[0]))
    assert t.column_names == ["metric_id", "tag_key", "tag_value", "tsid"]
    # writer sorts by (tag_key, tag_value, tsid)
    assert t.column("tag_key").to_pylist() == [b"a", b"a", b"b"]
    assert t.column("tag_value").to_pylist() == [b"x", b"z", b"y"]
    assert t.column("tsid").to_pylist() == [3, 1, 2]


def test_oracle_index_query_semantics(tmp_path):
    # oracle postings: union across index SSTs, AND=intersection, OR=union
    from horaedb_amd import Store
    from oracle.tag_index import index_query
    d, m, ids = _mk_store(tmp_path)
    with Store(d) as st:
        st.index_write(["dc", "dc", "env"], ["a", "a", "p"],
                       np.array([5, 7, 5], dtype=np.uint64))
        st.index_write(["dc"], ["a"], np.array([6], dtype=np.uint64))
    assert index_query(d, [("dc", "a")]).tolist() == [5, 6, 7]
    assert index_query(d, [("dc", "a"), ("env", "p")],
                       combine="and").tolist() == [5]
    assert index_query(d, [("dc", "a"), ("env", "p")],
                       combine="or").tolist() == [5, 6, 7]
    assert index_query(d, [("dc", "zz")]).tolist() == []
