# oracle/scan.py — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py header).
#
# CPU (numpy) restatement of the reference scan hot path:
#   decode -> filter -> sort-preserving merge -> MergeExec dedup -> aggregate.
# Each function cites the reference code it follows (paths relative to
# /root/reference/src/columnar_storage/src unless noted).
"""Restated semantics, in reference terms:

* An SST's rows are sorted by (pk columns..., ) with one ``__seq__`` per file
  (storage.rs:189-225: file id == sequence; fill_builtin_columns
  types.rs:219-239). Test scaffolding may use per-row seq (read.rs:512-573);
  we accept a per-row seq array.
* The scan plan (read.rs:429-494) is, per time segment:
  ParquetExec (decode) -> FilterExec (row predicates) ->
  SortPreservingMergeExec on [pk... ASC, __seq__ ASC] ->
  MergeExec (read.rs:262-343): group adjacent equal-PK runs of the merged
  stream, apply MergeOperator per group, strip builtin columns.
  Segments are unioned (storage.rs:343-368) in ascending time order.
* LastValueOperator (operator.rs:37-44): keep the LAST row of the group.
* BytesMergeOperator (operator.rs:47-111): concatenate the binary value
  column over the group; all other columns take the FIRST row.
"""
import numpy as np

MERGE_LAST = "last"      # UpdateMode::Overwrite -> LastValueOperator (read.rs:482-492)
MERGE_APPEND = "append"  # UpdateMode::Append   -> BytesMergeOperator


class SstBatch:
    """One SST's content (or one pre-sorted stream batch in test scaffolding).

    cols: list of 1-D numpy arrays, layout = [pk0..pk{k-1}, value..., ]
          (builtins NOT included; seq passed separately).
    seq:  scalar (per-file sequence, the production case) or per-row array.
    Rows must be sorted by (pk..., seq) — the SST writer invariant
    (storage.rs:244-256 sort_batch + SortPreservingMerge input contract).
    """

    def __init__(self, cols, seq):
        self.cols = [np.asarray(c) if not isinstance(c, np.ndarray) or c.dtype != object else c
                     for c in cols]
        n = len(self.cols[0]) if self.cols else 0
        if np.isscalar(seq):
            self.seq = np.full(n, seq, dtype=np.uint64)
        else:
            self.seq = np.asarray(seq, dtype=np.uint64)

    @property
    def n_rows(self):
        return len(self.cols[0]) if self.cols else 0


def _lex_order(key_arrays):
    """Stable ascending sort order with significance key_arrays[0] first.

    np.lexsort sorts by the LAST key most significantly and is stable —
    mirrors SortPreservingMergeExec's [pk... ASC, __seq__ ASC] with input
    (stream) order preserved for full ties (read.rs:412-427, :479-480).
    object (bytes) keys are not hashed: converted via ordered factorization.
    """
    keys = []
    for a in reversed(key_arrays):
        if a.dtype == object:  # binary PK column (read.rs:278-287)
            uniq = sorted(set(a.tolist()))
            rank = {v: i for i, v in enumerate(uniq)}
            a = np.fromiter((rank[v] for v in a.tolist()), dtype=np.int64, count=len(a))
        keys.append(a)
    return np.lexsort(keys)


def _pk_run_bounds(pk_cols):
    """Start indices of adjacent equal-PK runs (MergeExec grouping,
    read.rs:289-306: scalar loop comparing row i with row i+1 per PK col)."""
    n = len(pk_cols[0])
    if n == 0:
        return np.empty(0, dtype=np.int64)
    diff = np.zeros(n - 1, dtype=bool)
    for c in pk_cols:
        if c.dtype == object:
            d = np.fromiter((c[i] != c[i + 1] for i in range(n - 1)), dtype=bool, count=n - 1)
        else:
            d = c[:-1] != c[1:]
        diff |= d
    return np.concatenate(([0], np.nonzero(diff)[0] + 1))


def merge_scan(ssts, num_primary_keys, merge_op=MERGE_LAST, value_idxes=None,
               predicate=None, keep_builtin=False):
    """Merged, deduplicated scan of one time segment.

    ssts: list of SstBatch (the segment's SSTs / streams).
    predicate: f(cols)->bool mask, applied per SST BEFORE the merge —
        FilterExec sits under SortPreservingMergeExec (read.rs:456-480).
    Returns list of columns: user columns (pk + values); builtins stripped
        (read.rs:330-343) unless keep_builtin (compaction mode).
    Output rows are the segment stream order: sorted by (pk..., seq).
    """
    cols_cat, seq_cat = None, []
    for sst in ssts:
        cols = sst.cols
        seq = sst.seq
        if predicate is not None:
            mask = predicate(cols)
            cols = [c[mask] for c in cols]
            seq = seq[mask]
        if cols_cat is None:
            cols_cat = [[] for _ in cols]
        for i, c in enumerate(cols):
            cols_cat[i].append(c)
        seq_cat.append(seq)
    if cols_cat is None:
        return []
    cols = [np.concatenate(parts) if parts[0].dtype != object
            else np.array(sum((p.tolist() for p in parts), []), dtype=object)
            for parts in cols_cat]
    seq = np.concatenate(seq_cat)
    n = len(seq)
    if n == 0:
        return [c[:0] for c in cols] + ([seq[:0]] if keep_builtin else [])

    order = _lex_order([*cols[:num_primary_keys], seq])
    cols = [c[order] for c in cols]
    seq = seq[order]

    starts = _pk_run_bounds(cols[:num_primary_keys])
    ends = np.concatenate((starts[1:], [n]))

    if merge_op == MERGE_LAST:
        take = ends - 1  # LastValueOperator: batch.slice(num_rows-1, 1) operator.rs:40-43
        out = [c[take] for c in cols]
        out_seq = seq[take]
    elif merge_op == MERGE_APPEND:
        if value_idxes is None:
            raise ValueError("MERGE_APPEND needs value_idxes (BytesMergeOperator::new)")
        out = []
        for idx, c in enumerate(cols):
            if idx in value_idxes:
                # concatenate all elements of the group (operator.rs:76-98)
                merged = np.array(
                    [b"".join(c[s:e].tolist()) for s, e in zip(starts, ends)],
                    dtype=object)
                out.append(merged)
            else:
                out.append(c[starts])  # take first (operator.rs:99-102)
        out_seq = seq[starts]
    else:
        raise ValueError(merge_op)

    if keep_builtin:
        return out + [out_seq]
    return out


# ---------------------------------------------------------------------------
# Metric-shaped scan+aggregate (the north-star query; aggregate semantics are
# pinned by BASELINE.json configs + rfc:218-231 — no reference code exists,
# SURVEY §2 fact 3).
# ---------------------------------------------------------------------------

AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, AGG_AVG = 1, 2, 4, 8, 16


def scan_agg(ssts, ts_range, series_set=None, bucket_ms=0,
             ops=AGG_SUM | AGG_COUNT):
    """range+tag filter -> dedup merge -> group-by aggregate.

    ssts: list of SstBatch with cols = [series u64, ts i64, value f64]
          (the metric shape, rfc:218-231; schema contract types.rs:150-240).
          May span several time segments — since PK includes ts and segments
          partition ts, merging all SSTs at once is equivalent to the
          reference's per-segment merge + ascending-segment union.
    ts_range: (start, end) half-open [start, end) — types.rs:46-133.
    series_set: optional iterable of series ids (tag predicate materialized
          as id-set membership, BASELINE config 3).
    Returns dict of 1-D arrays sorted by (series_id[, bucket]):
          series_id, bucket?, sum, count, vmin, vmax, avg (requested ops).
    Sums accumulate in ts order within each group (deterministic reference
    order for the 1e-9-relative GPU comparison).
    """
    lo, hi = ts_range
    sset = None
    if series_set is not None:
        sset = np.asarray(sorted(set(int(s) for s in series_set)), dtype=np.uint64)

    def pred(cols):
        s, t = cols[0], cols[1]
        m = (t >= lo) & (t < hi)
        if sset is not None:
            m &= np.isin(s, sset)
        return m

    merged = merge_scan(ssts, num_primary_keys=2, merge_op=MERGE_LAST,
                        predicate=pred)
    if not merged or len(merged[0]) == 0:
        res = {"series_id": np.empty(0, np.uint64)}
        if bucket_ms:
            res["bucket"] = np.empty(0, np.int64)
        for name in ("sum", "vmin", "vmax", "avg"):
            res[name] = np.empty(0, np.float64)
        res["count"] = np.empty(0, np.uint64)
        return res

    s, t, v = merged[0], merged[1], merged[2].astype(np.float64)
    if bucket_ms:
        b = np.floor_divide(t, np.int64(bucket_ms))  # truncate_by types.rs:82-86
        # merged stream is sorted by (series, ts) => (series, bucket) runs are adjacent
        kstart = np.concatenate(([0], np.nonzero((s[:-1] != s[1:]) | (b[:-1] != b[1:]))[0] + 1))
    else:
        b = None
        kstart = np.concatenate(([0], np.nonzero(s[:-1] != s[1:])[0] + 1))
    kend = np.concatenate((kstart[1:], [len(s)]))

    res = {"series_id": s[kstart]}
    if bucket_ms:
        res["bucket"] = b[kstart]
    if ops & (AGG_SUM | AGG_AVG):
        res["sum"] = np.add.reduceat(v, kstart)
    if ops & (AGG_COUNT | AGG_AVG):
        res["count"] = (kend - kstart).astype(np.uint64)
    if ops & AGG_MIN:
        res["vmin"] = np.minimum.reduceat(v, kstart)
    if ops & AGG_MAX:
        res["vmax"] = np.maximum.reduceat(v, kstart)
    if ops & AGG_AVG:
        res["avg"] = res["sum"] / res["count"]
    if not (ops & AGG_SUM):
        res.pop("sum", None)
    if not (ops & AGG_COUNT):
        res.pop("count", None)
    return res


# ---------------------------------------------------------------------------
# SST reading (decode oracle = pyarrow) and schema helpers.
# ---------------------------------------------------------------------------

def read_sst(path, with_builtins=False):
    """Read a metric-shaped SST written by the reference writer contract
    (storage.rs:193-213; schema types.rs:150-240). Returns SstBatch with
    cols [series u64, ts i64, value f64]; seq is read from the __seq__
    column (constant per file) falling back to the numeric file stem
    (file id == sequence, sst.rs:39-46,:193-205)."""
    import pyarrow.parquet as pq
    import os
    t = pq.read_table(path)
    names = t.schema.names
    need = ["series_id", "timestamp", "value"]
    for nm in need:
        if nm not in names:
            raise ValueError(f"{path}: missing column {nm} (schema contract)")
    s = t.column("series_id").to_numpy().astype(np.uint64)
    ts = t.column("timestamp").to_numpy().astype(np.int64)
    import pyarrow as pa
    vcol = t.column("value")
    if pa.types.is_binary(vcol.type) or pa.types.is_large_binary(vcol.type) \
            or pa.types.is_string(vcol.type):
        # Binary value schema (BytesMergeOperator stores, operator.rs:47-111)
        v = np.empty(t.num_rows, dtype=object)
        for i, x in enumerate(vcol.to_pylist()):
            v[i] = x if isinstance(x, bytes) else x.encode()
    else:
        v = vcol.to_numpy().astype(np.float64)
    if "__seq__" in names and t.num_rows > 0:
        sq = t.column("__seq__").to_numpy().astype(np.uint64)
        if (sq == sq[0]).all():
            seq = int(sq[0])
        else:
            # keep_builtin compaction output (executor.rs:155-222): the
            # file carries PER-ROW sequences; MergeStream orders by the
            # row's own __seq__ (read.rs:289-343)
            seq = sq
    else:
        seq = int(os.path.splitext(os.path.basename(path))[0])
    cols = [s, ts, v]
    if with_builtins:
        cols += [t.column("__seq__").to_numpy().astype(np.uint64),
                 np.zeros(t.num_rows, np.uint64)]
    return SstBatch(cols, seq)


def truncate_by(ts, duration_ms):
    """Timestamp::truncate_by (types.rs:82-86): floor to the duration."""
    return (int(ts) // int(duration_ms)) * int(duration_ms)


def fill_required_projections(projection, num_primary_keys, seq_idx):
    """StorageSchema::fill_required_projections (types.rs:203-216):
    force PKs then __seq__ into a Some(projection), preserving order of the
    caller's entries; None stays None."""
    if projection is None:
        return None
    proj = list(projection)
    for i in range(num_primary_keys):
        if i not in proj:
            proj.append(i)
    if seq_idx not in proj:
        proj.append(seq_idx)
    return proj


def scan_rows(ssts, ts_range, series_set=None, segment_ms=12 * 3600 * 1000):
    """The merged, deduplicated row STREAM (hx_scan parity): segments in
    ascending time order (union of per-segment plans, storage.rs:343-368),
    rows sorted by (series_id, timestamp) within each segment."""
    lo, hi = ts_range
    sset = None
    if series_set is not None:
        sset = np.asarray(sorted(set(int(x) for x in series_set)), np.uint64)

    def pred(cols):
        m = (cols[1] >= lo) & (cols[1] < hi)
        if sset is not None:
            m &= np.isin(cols[0], sset)
        return m

    merged = merge_scan(ssts, num_primary_keys=2, merge_op=MERGE_LAST,
                        predicate=pred)
    if not merged or len(merged[0]) == 0:
        return {"series_id": np.empty(0, np.uint64),
                "timestamp": np.empty(0, np.int64),
                "value": np.empty(0, np.float64)}
    s, t, v = merged[0], merged[1], merged[2]
    seg = np.floor_divide(t, np.int64(segment_ms))
    order = np.lexsort((t, s, seg))
    return {"series_id": s[order], "timestamp": t[order], "value": v[order]}
