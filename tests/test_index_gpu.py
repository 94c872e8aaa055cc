# Inverted-index parity (SURVEY §8 f2 / VERDICT r01 missing #1): the GPU
# tag→TSID query (hx_index_query: BYTE_ARRAY decode + postings filter +
# set combine) against the oracle restatement of rfc:86-137, and the full
# rfc query path: tag filter -> TSID set -> data scan == oracle scan with
# the same series set.
import os
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

pytestmark = pytest.mark.gpu


def _mk_store(tmp_path, n_series=500, n_rows=50_000):
    from tools.gen_ssts import gen_dataset, middle_range
    d = str(tmp_path / "store")
    m = gen_dataset(d, n_rows=n_rows, n_series=n_series, n_ssts=4, seed=21)
    ids = np.load(os.path.join(d, "series_ids.npy"))
    return d, m, ids


def _write_index(store, ids, rng):
    # RFC-style tags: host=h{i} (unique), dc=dc{i%10} (10% selectivity),
    # env in {prod, dev} (50/50)
    n = len(ids)
    keys, vals, tsids = [], [], []
    for i, t in enumerate(ids):
        keys += ["host", "dc", "env"]
        vals += [f"h{i}", f"dc{i % 10}", "prod" if i % 2 == 0 else "dev"]
        tsids += [t, t, t]
    # write in two SSTs with a shuffled split (query must union postings
    # across index files and the writer must sort rows)
    order = rng.permutation(len(tsids))
    half = len(order) // 2
    for part in (order[:half], order[half:]):
        store.index_write([keys[i] for i in part],
                          [vals[i] for i in part],
                          np.array([tsids[i] for i in part], dtype=np.uint64))


def test_index_query_parity(tmp_path):
    from horaedb_amd import Store
    from oracle.tag_index import index_query
    d, m, ids = _mk_store(tmp_path)
    rng = np.random.default_rng(3)
    with Store(d) as st:
        _write_index(st, ids, rng)
        for preds, combine in [
            ([("dc", "dc3")], "and"),
            ([("env", "prod")], "and"),
            ([("dc", "dc3"), ("env", "prod")], "and"),
            ([("dc", "dc3"), ("dc", "dc4")], "or"),
            ([("host", "h7")], "and"),
            ([("dc", "nope")], "and"),
            ([("dc", "dc1"), ("host", "h999999")], "and"),
        ]:
            got = st.index_query(preds, combine=combine)
            exp = index_query(d, preds, combine=combine)
            np.testing.assert_array_equal(got, exp), (preds, combine)


def test_index_to_scan_path(tmp_path):
    # the rfc's full query path: label filter -> TSID set -> data scan
    import oracle
    from oracle.scan import AGG_SUM, AGG_COUNT
    from oracle.tag_index import index_query
    from horaedb_amd import Store, AGG_SUM as HSUM, AGG_COUNT as HCNT
    from tools.gen_ssts import middle_range

    d, m, ids = _mk_store(tmp_path)
    rng = np.random.default_rng(4)
    lo, hi = middle_range(m)
    with Store(d) as st:
        _write_index(st, ids, rng)
        tsids = st.index_query([("dc", "dc5"), ("env", "dev")],
                               combine="and")
        res = st.scan_agg((lo, hi), ops=HSUM | HCNT, devices=[0],
                          series_in=tsids.tolist())
    exp_ids = index_query(d, [("dc", "dc5"), ("env", "dev")], combine="and")
    np.testing.assert_array_equal(tsids, exp_ids)
    ssts = [oracle.read_sst(s["path"]) for s in m["ssts"]]
    exp = oracle.scan_agg(ssts, (lo, hi), series_set=set(exp_ids.tolist()),
                          ops=AGG_SUM | AGG_COUNT)
    assert res["series_id"].tolist() == exp["series_id"].tolist()
    np.testing.assert_array_equal(res["count"], exp["count"])
    np.testing.assert_allclose(res["sum"], exp["sum"], rtol=1e-9)
