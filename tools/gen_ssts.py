#!/usr/bin/env python3
# tools/gen_ssts.py — synthetic SST generator (shared by tests and bench.py).
#
# Writes metric-shaped Parquet SSTs exactly as the reference writer contract
# produces them (storage.rs:193-213; defaults config.rs:120-133: row-group
# 8192, PLAIN, dict off; encodings enum config.rs:54-75; Snappy or
# uncompressed per config):
#   (series_id u64 PK, timestamp i64 PK, value f64, __seq__ u64, __reserved__ u64)
#   rows sorted by (series_id, timestamp); one __seq__ per file = file id;
#   file path {out}/data/{seq}.sst (sst.rs:193-205).
#
# Dataset shape per BASELINE.json / SURVEY §8(d): n_series ids (sorted unique
# u64), n_points = n_rows/n_series timestamps per series at 10s steps from
# ts_start; SST w holds the w-th contiguous time-index window (disjoint ts
# ranges => no cross-SST duplicate PKs, the compacted layout; overlap
# scenarios for dedup tests are built with gen_sst_from_arrays).
#
# value(s_idx, t) is a counter-based splitmix64 hash -> U[0,1) so any row's
# value is reproducible independent of chunking or worker count.
import argparse
import os
import json
import numpy as np

TS_START_DEFAULT = 1735689600000  # 2025-01-01T00:00:00Z, SURVEY §8(d)
STEP_MS_DEFAULT = 10_000

SCHEMA_COLS = ["series_id", "timestamp", "value", "__seq__", "__reserved__"]


def _pa_schema():
    import pyarrow as pa
    return pa.schema([
        pa.field("series_id", pa.uint64(), nullable=False),
        pa.field("timestamp", pa.int64(), nullable=False),
        pa.field("value", pa.float64(), nullable=False),
        pa.field("__seq__", pa.uint64(), nullable=False),
        pa.field("__reserved__", pa.uint64(), nullable=False),
    ])


def splitmix64(x):
    """Vectorized splitmix64 on uint64 arrays (wrapping arithmetic)."""
    x = x.astype(np.uint64, copy=True)
    x += np.uint64(0x9E3779B97F4A7C15)
    z = x
    z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
    z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    return z ^ (z >> np.uint64(31))


def value_of(seed, global_idx):
    """value = U[0,1) from splitmix64(seed*GOLDEN ^ global_idx)."""
    mixed_seed = np.uint64((int(seed) * 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF)
    h = splitmix64(mixed_seed ^ global_idx.astype(np.uint64))
    return (h >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)


def make_series_ids(n_series, seed):
    """n_series sorted unique u64 ids (seahash-of-labels style)."""
    rng = np.random.default_rng(np.random.PCG64(seed))
    ids = rng.integers(0, np.iinfo(np.uint64).max, n_series, dtype=np.uint64)
    ids = np.unique(ids)
    while len(ids) < n_series:  # collision top-up (vanishingly rare)
        extra = rng.integers(0, np.iinfo(np.uint64).max,
                             n_series - len(ids) + 16, dtype=np.uint64)
        ids = np.unique(np.concatenate([ids, extra]))
    return ids[:n_series]


def write_sst(path, series, ts, value, seq, row_group=8192,
              compression="none", ts_encoding="PLAIN", dict_columns=(),
              slim_builtins=False, data_page_version="1.0"):
    """Write one SST from explicit row arrays (must be (series,ts)-sorted)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    n = len(series)
    tbl = pa.table({
        "series_id": np.ascontiguousarray(series, dtype=np.uint64),
        "timestamp": np.ascontiguousarray(ts, dtype=np.int64),
        "value": np.ascontiguousarray(value, dtype=np.float64),
        "__seq__": np.full(n, seq, dtype=np.uint64),
        "__reserved__": np.zeros(n, dtype=np.uint64),
    }, schema=_pa_schema())
    kw = {}
    if slim_builtins and not dict_columns:
        # __seq__/__reserved__ are constant per file and never read by the
        # scan (seq comes from the file id); dictionary-encode them to cut
        # ~40% of file size for the 8-GPU bench datasets. The three DATA
        # columns keep their configured encodings.
        dict_columns = ("__seq__", "__reserved__")
    if dict_columns:
        # RLE_DICTIONARY chunks (config.rs:54-75 with dictionaries on);
        # large limits keep one dictionary per chunk (no PLAIN fallback)
        kw["use_dictionary"] = list(dict_columns)
        kw["dictionary_pagesize_limit"] = 1 << 30
        enc = {c: "PLAIN" for c in SCHEMA_COLS if c not in dict_columns}
        enc["timestamp"] = ts_encoding if "timestamp" not in dict_columns \
            else None
        enc = {c: v for c, v in enc.items() if v}
    else:
        kw["use_dictionary"] = False
        enc = {c: "PLAIN" for c in SCHEMA_COLS}
        enc["timestamp"] = ts_encoding
    pq.write_table(
        tbl, path, row_group_size=row_group,
        compression="NONE" if compression == "none" else compression.upper(),
        data_page_version=data_page_version, column_encoding=enc,
        write_statistics=True, **kw)
    return n


def gen_sst_from_arrays(store_dir, seq, series, ts, value, sort=True, **kw):
    """Test scenario helper: build {store}/data/{seq}.sst from explicit rows.
    sort=True applies the writer's (pk...)-stable sort (storage.rs:244-256)."""
    series = np.asarray(series, dtype=np.uint64)
    ts = np.asarray(ts, dtype=np.int64)
    value = np.asarray(value, dtype=np.float64)
    if sort:
        order = np.lexsort((ts, series))  # stable: equal PKs keep input order
        series, ts, value = series[order], ts[order], value[order]
    ddir = os.path.join(store_dir, "data")
    os.makedirs(ddir, exist_ok=True)
    path = os.path.join(ddir, f"{seq}.sst")
    write_sst(path, series, ts, value, seq, **kw)
    return path


def _gen_one(args):
    (j, out_dir, ids_path, n_series, n_points, n_ssts, seed, ts_start,
     step_ms, row_group, compression, ts_encoding) = args
    # j enumerates (generation, window): generation g re-writes the SAME
    # (series, ts) PKs of window w with new values and a higher seq — the
    # ts-overlap layout (overwrite churn before compaction) that exercises
    # the cross-SST MergeExec dedup (read.rs:100-391) at scale. g = 0 is the
    # plain disjoint layout.
    w, g = j % n_ssts, j // n_ssts
    ids = np.load(ids_path, mmap_mode="r")
    t0 = w * n_points // n_ssts
    t1 = (w + 1) * n_points // n_ssts
    k = t1 - t0
    if k == 0:
        return None
    series = np.repeat(ids, k)
    t_idx = np.tile(np.arange(t0, t1, dtype=np.int64), n_series)
    ts = ts_start + t_idx * step_ms
    gidx = (np.repeat(np.arange(n_series, dtype=np.uint64), k) *
            np.uint64(n_points)) + t_idx.astype(np.uint64)
    value = value_of(seed + 7919 * g, gidx)
    seq = j + 1
    path = os.path.join(out_dir, "data", f"{seq}.sst")
    n = write_sst(path, series, ts, value, seq, row_group=row_group,
                  compression=compression, ts_encoding=ts_encoding,
                  slim_builtins=True)
    return {"seq": seq, "path": path, "rows": int(n),
            "ts_min": int(ts_start + t0 * step_ms),
            "ts_max": int(ts_start + (t1 - 1) * step_ms)}


def gen_dataset(out_dir, n_rows, n_series, n_ssts, seed=42,
                ts_start=TS_START_DEFAULT, step_ms=STEP_MS_DEFAULT,
                row_group=8192, compression="none", ts_encoding="PLAIN",
                workers=1, overlap_gens=1):
    """overlap_gens > 1: each further generation re-writes the SAME PKs with
    new values under higher seqs (ts-overlapping SSTs) — the pre-compaction
    overwrite-churn layout; total rows = n_rows * overlap_gens, surviving
    rows after dedup = the newest generation only."""
    assert n_rows % n_series == 0, "n_rows must be a multiple of n_series"
    n_points = n_rows // n_series
    os.makedirs(os.path.join(out_dir, "data"), exist_ok=True)
    ids = make_series_ids(n_series, seed)
    ids_path = os.path.join(out_dir, "series_ids.npy")
    np.save(ids_path, ids)
    jobs = [(j, out_dir, ids_path, n_series, n_points, n_ssts, seed,
             ts_start, step_ms, row_group, compression, ts_encoding)
            for j in range(n_ssts * max(1, overlap_gens))]
    if workers > 1:
        import multiprocessing as mp
        with mp.get_context("spawn").Pool(workers) as pool:
            metas = pool.map(_gen_one, jobs)
    else:
        metas = [_gen_one(j) for j in jobs]
    metas = [m for m in metas if m]
    manifest = {
        "n_rows": n_rows * max(1, overlap_gens), "n_series": n_series,
        "n_points": n_points, "overlap_gens": max(1, overlap_gens),
        "n_ssts": len(metas), "seed": seed, "ts_start": ts_start,
        "step_ms": step_ms, "row_group": row_group,
        "compression": compression, "ts_encoding": ts_encoding,
        "ts_end": ts_start + n_points * step_ms, "ssts": metas,
    }
    with open(os.path.join(out_dir, "dataset.json"), "w") as f:
        json.dump(manifest, f, indent=1)
    return manifest


def write_bytes_sst(path, series, ts, values, seq, row_group=8192,
                    sort=True):
    """One Binary-value SST (BytesMergeOperator stores, operator.rs:47-111):
    (series u64 PK, ts i64 PK, value binary, builtins), PLAIN uncompressed,
    writer-sorted by (series, ts) stable (equal PKs keep input order)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    series = np.asarray(series, dtype=np.uint64)
    ts = np.asarray(ts, dtype=np.int64)
    values = list(values)
    if sort:
        order = np.lexsort((ts, series))
        series = series[order]
        ts = ts[order]
        values = [values[i] for i in order]
    n = len(series)
    schema = pa.schema([
        pa.field("series_id", pa.uint64(), nullable=False),
        pa.field("timestamp", pa.int64(), nullable=False),
        pa.field("value", pa.binary(), nullable=False),
        pa.field("__seq__", pa.uint64(), nullable=False),
        pa.field("__reserved__", pa.uint64(), nullable=False),
    ])
    tbl = pa.table({
        "series_id": pa.array(series, pa.uint64()),
        "timestamp": pa.array(ts, pa.int64()),
        "value": pa.array([v if isinstance(v, bytes) else v.encode()
                           for v in values], pa.binary()),
        "__seq__": pa.array(np.full(n, seq, np.uint64), pa.uint64()),
        "__reserved__": pa.array(np.zeros(n, np.uint64), pa.uint64()),
    }, schema=schema)
    os.makedirs(os.path.dirname(path), exist_ok=True)
    pq.write_table(tbl, path, row_group_size=row_group, compression="NONE",
                   use_dictionary=False, data_page_version="1.0",
                   column_encoding={c: "PLAIN" for c in
                                    ("series_id", "timestamp", "value",
                                     "__seq__", "__reserved__")},
                   write_statistics=True)
    return path


def gen_tag_index(store_dir, n_dc=100):
    """Write {store}/index/1.sst (the RFC index table, rfc:86-137) for the
    dataset's series: tags dc=dc{i%n_dc} (1/n_dc selectivity each) and
    env in {prod, dev}. Vectorized pyarrow write (PLAIN uncompressed,
    row-group 8192, stats) matching the native reader's layout contract;
    rows sorted by (tag_key, tag_value, tsid)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    idx_path = os.path.join(store_dir, "index", "1.sst")
    if os.path.exists(idx_path):
        return idx_path
    os.makedirs(os.path.dirname(idx_path), exist_ok=True)
    ids = np.load(os.path.join(store_dir, "series_ids.npy"))
    n = len(ids)
    # dc block: values dc0..dc{n_dc-1} in STRING sort order, tsids ascending
    dc_strings = sorted(f"dc{k}" for k in range(n_dc))
    rank = np.empty(n_dc, np.int64)
    for r, sname in enumerate(dc_strings):
        rank[int(sname[2:])] = r
    codes = rank[np.arange(n) % n_dc]
    order = np.lexsort((ids, codes))   # ids already ascending per group
    dc_vals = np.array([f"dc{k}" for k in range(n_dc)], dtype=object)
    dc_value_col = dc_vals[np.arange(n) % n_dc][order]
    dc_tsid = ids[order]
    # env block: dev (odd i) then prod (even i), tsids ascending per value
    env_is_prod = (np.arange(n) % 2) == 0
    env_order = np.lexsort((ids, env_is_prod))  # dev(False=0) first? no:
    # lexsort ascending: False(0) < True(1) -> dev rows first... but string
    # order is "dev" < "prod" and dev rows are the odd ones (is_prod False)
    env_value_col = np.where(env_is_prod[env_order], "prod", "dev")
    env_tsid = ids[env_order]
    schema = pa.schema([
        pa.field("metric_id", pa.uint64(), nullable=False),
        pa.field("tag_key", pa.binary(), nullable=False),
        pa.field("tag_value", pa.binary(), nullable=False),
        pa.field("tsid", pa.uint64(), nullable=False),
    ])   # REQUIRED columns: v1 pages carry no def-level prefix
    tbl = pa.table({
        "metric_id": pa.array(np.zeros(2 * n, np.uint64), pa.uint64()),
        "tag_key": pa.array([b"dc"] * n + [b"env"] * n, pa.binary()),
        "tag_value": pa.array(
            [x.encode() for x in dc_value_col] +
            [x.encode() for x in env_value_col], pa.binary()),
        "tsid": pa.array(np.concatenate([dc_tsid, env_tsid]), pa.uint64()),
    }, schema=schema)
    pq.write_table(tbl, idx_path, row_group_size=8192, compression="NONE",
                   use_dictionary=False, data_page_version="1.0",
                   column_encoding={c: "PLAIN" for c in
                                    ("metric_id", "tag_key", "tag_value",
                                     "tsid")},
                   write_statistics=True)
    return idx_path


def middle_range(manifest, frac=0.5):
    """The benchmark's ts-range: middle `frac` of the dataset span."""
    span = manifest["ts_end"] - manifest["ts_start"]
    lo = manifest["ts_start"] + int(span * (0.5 - frac / 2))
    hi = manifest["ts_start"] + int(span * (0.5 + frac / 2))
    return lo, hi


def main():
    p = argparse.ArgumentParser()
    p.add_argument("out_dir")
    p.add_argument("--rows", type=int, default=10_000_000)
    p.add_argument("--series", type=int, default=100_000)
    p.add_argument("--ssts", type=int, default=1)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--compression", default="none", choices=["none", "snappy", "zstd"])
    p.add_argument("--ts-encoding", default="PLAIN",
                   choices=["PLAIN", "DELTA_BINARY_PACKED"])
    p.add_argument("--workers", type=int, default=1)
    args = p.parse_args()
    m = gen_dataset(args.out_dir, args.rows, args.series, args.ssts,
                    seed=args.seed, compression=args.compression,
                    ts_encoding=args.ts_encoding, workers=args.workers)
    print(json.dumps({k: v for k, v in m.items() if k != "ssts"}))


if __name__ == "__main__":
    main()
