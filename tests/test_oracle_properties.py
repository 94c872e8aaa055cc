# Property-based hardening of the oracle itself (hypothesis): the numpy
# restatement must agree with a direct, row-at-a-time model of the reference
# semantics (last-seq-wins dedup over (series, ts), filter before merge)
# on arbitrary small inputs. Strengthens DESIGN.md §7's oracle pinning.
import numpy as np
from hypothesis import given, settings, strategies as st

import oracle
from oracle import SstBatch
from oracle.scan import AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, AGG_AVG

OPS_ALL = AGG_SUM | AGG_COUNT | AGG_MIN | AGG_MAX | AGG_AVG


def brute_force(sst_rows, ts_range, bucket_ms=0, series_set=None):
    """Row-at-a-time model: for each (series, ts) PK the surviving row is
    the one from the highest seq; among equal (seq) the later row in that
    SST's order (LastValueOperator). Filter applies per row before dedup
    (equal PKs share ts, hence share the range-filter outcome)."""
    winner = {}  # (series, ts) -> (seq, ordinal, value)
    for seq, rows in sst_rows:
        for i, (s, t, v) in enumerate(rows):
            if not (ts_range[0] <= t < ts_range[1]):
                continue
            if series_set is not None and s not in series_set:
                continue
            key = (s, t)
            cur = winner.get(key)
            if cur is None or (seq, i) > (cur[0], cur[1]):
                winner[key] = (seq, i, v)
    groups = {}
    for (s, t), (_, _, v) in winner.items():
        gk = (s, t // bucket_ms) if bucket_ms else (s,)
        groups.setdefault(gk, []).append((t, v))
    out = {}
    for gk, tv in groups.items():
        tv.sort()
        vals = [v for _, v in tv]
        out[gk] = {"sum": sum(vals), "count": len(vals),
                   "vmin": min(vals), "vmax": max(vals)}
    return out


row_strategy = st.tuples(
    st.integers(0, 6),                      # series
    st.integers(-20, 20),                   # ts (exercises negatives)
    st.integers(-1000, 1000),               # value (int-valued f64: exact)
)

sst_strategy = st.lists(st.lists(row_strategy, min_size=0, max_size=30),
                        min_size=1, max_size=5)


def build_ssts(raw):
    ssts = []
    for fid, rows in enumerate(raw):
        arr = sorted(rows)  # (series, ts) PK sort; stable keeps batch order
        s = np.array([r[0] for r in arr], np.uint64)
        t = np.array([r[1] for r in arr], np.int64)
        v = np.array([float(r[2]) for r in arr], np.float64)
        ssts.append(SstBatch([s, t, v], fid + 1))
    return ssts, [(fid + 1, sorted(rows)) for fid, rows in enumerate(raw)]


@settings(max_examples=120, deadline=None)
@given(raw=sst_strategy, lo=st.integers(-15, 15), width=st.integers(0, 25),
       bucket=st.sampled_from([0, 3, 7]),
       sel=st.one_of(st.none(), st.sets(st.integers(0, 6), max_size=4)))
def test_scan_agg_matches_brute_force(raw, lo, width, bucket, sel):
    ssts, sst_rows = build_ssts(raw)
    tr = (lo, lo + width)
    res = oracle.scan_agg(ssts, tr, bucket_ms=bucket, ops=OPS_ALL,
                          series_set=sel)
    exp = brute_force(sst_rows, tr, bucket_ms=bucket, series_set=sel)
    keys = list(zip(res["series_id"].tolist(),
                    res["bucket"].tolist())) if bucket else \
        [(s,) for s in res["series_id"].tolist()]
    assert keys == sorted(exp.keys())
    for i, k in enumerate(keys):
        e = exp[k]
        assert res["count"][i] == e["count"]
        assert res["vmin"][i] == e["vmin"]
        assert res["vmax"][i] == e["vmax"]
        assert res["sum"][i] == e["sum"]  # int-valued f64: exact
        assert res["avg"][i] == e["sum"] / e["count"]


@settings(max_examples=80, deadline=None)
@given(raw=sst_strategy, lo=st.integers(-15, 15), width=st.integers(0, 25),
       sel=st.sets(st.integers(0, 6), max_size=4))
def test_scan_rows_matches_brute_force(raw, lo, width, sel):
    ssts, sst_rows = build_ssts(raw)
    tr = (lo, lo + width)
    sset = sel if sel else None
    res = oracle.scan_rows(ssts, tr, series_set=sset, segment_ms=10)
    # scan_rows: one surviving row per PK, ordered by (segment, series, ts)
    # with floor-division segments (negative ts => negative segments)
    winners = brute_force_winners(sst_rows, tr, sset)
    want = sorted((t // 10, s, t, v)
                  for (s, t), (_, _, v) in winners.items())
    assert res["series_id"].tolist() == [s for _, s, _, _ in want]
    assert res["timestamp"].tolist() == [t for _, _, t, _ in want]
    assert res["value"].tolist() == [v for _, _, _, v in want]


def brute_force_winners(sst_rows, ts_range, series_set=None):
    winner = {}
    for seq, rows in sst_rows:
        for i, (s, t, v) in enumerate(rows):
            if not (ts_range[0] <= t < ts_range[1]):
                continue
            if series_set is not None and s not in series_set:
                continue
            cur = winner.get((s, t))
            if cur is None or (seq, i) > (cur[0], cur[1]):
                winner[(s, t)] = (seq, i, float(v))
    return winner


@settings(max_examples=60, deadline=None)
@given(raw=st.lists(st.lists(st.tuples(
           st.integers(0, 4),          # series
           st.integers(0, 10),         # ts
           st.integers(-99, 99),       # value
           st.integers(1, 5)),         # per-row seq (test scaffolding)
       min_size=0, max_size=20), min_size=1, max_size=3))
def test_merge_scan_per_row_seq(raw):
    # oracle's per-row __seq__ support (the reference's own test scaffolding
    # uses per-row seq, read.rs:512-573): winner = max (seq, stream order)
    from oracle.scan import merge_scan, MERGE_LAST
    ssts = []
    model_rows = []
    for batch in raw:
        # stream batches must be (pk..., seq)-sorted (merge input contract)
        arr = sorted(batch, key=lambda r: (r[0], r[1], r[3]))
        s = np.array([r[0] for r in arr], np.uint64)
        t = np.array([r[1] for r in arr], np.int64)
        v = np.array([float(r[2]) for r in arr], np.float64)
        q = np.array([r[3] for r in arr], np.uint64)
        ssts.append(SstBatch([s, t, v], q))
        model_rows.append(arr)
    out = merge_scan(ssts, num_primary_keys=2)
    winner = {}
    for bi, rows in enumerate(model_rows):
        for i, (s, t, v, q) in enumerate(rows):
            cur = winner.get((s, t))
            # lexsort ties (same pk, same seq) resolve by stream order:
            # batch index then row index
            if cur is None or (q, bi, i) >= (cur[0], cur[1], cur[2]):
                winner[(s, t)] = (q, bi, i, float(v))
    want = sorted(((s, t, v) for (s, t), (_, _, _, v) in winner.items()))
    got = list(zip(out[0].tolist(), out[1].tolist(), out[2].tolist()))
    assert got == want
