# oracle/tag_index.py — TEST INFRASTRUCTURE ONLY.
#
# CPU restatement (pyarrow + numpy) of the inverted-index query semantics
# the product's hx_index_query implements on GPU. Pinned by the reference
# RFC (docs/rfcs/20240827-metric-engine.md:86-137): the `index` table holds
# rows (MetricID u64, TagKey bytes, TagValue bytes, TSID u64), PK-sorted by
# (tag_key, tag_value, tsid); a label filter `k=v` resolves to the TSID
# postings of (k, v) unioned across index SSTs; multiple filters intersect
# (AND) or union (OR). (The reference's own index module is an uncompiled
# skeleton — metric_engine/src/index/mod.rs, SURVEY §2 — so the RFC text is
# the semantic anchor, like VictoriaMetrics' label=value -> metricID
# inverted index it cites.)
import glob
import os

import numpy as np


def read_index(store_dir):
    """All index rows of {store}/index/*.sst as numpy arrays."""
    import pyarrow.parquet as pq
    keys, vals, tsids = [], [], []
    for p in sorted(glob.glob(os.path.join(store_dir, "index", "*.sst"))):
        t = pq.read_table(p)
        keys.append(t.column("tag_key").to_numpy(zero_copy_only=False))
        vals.append(t.column("tag_value").to_numpy(zero_copy_only=False))
        tsids.append(t.column("tsid").to_numpy(zero_copy_only=False)
                     .astype(np.uint64))
    if not keys:
        return (np.empty(0, object), np.empty(0, object),
                np.empty(0, np.uint64))
    return (np.concatenate(keys), np.concatenate(vals),
            np.concatenate(tsids))


def index_query(store_dir, preds, combine="and"):
    """preds: [(key, value), ...] -> sorted distinct TSID set (u64)."""
    keys, vals, tsids = read_index(store_dir)
    sets = []
    for k, v in preds:
        # the index stores raw bytes (RFC: TagKey/TagValue are `bytes`)
        if isinstance(k, str):
            k = k.encode()
        if isinstance(v, str):
            v = v.encode()
        mask = (keys == k) & (vals == v)
        sets.append(np.unique(tsids[mask]))
    if not sets:
        return np.empty(0, np.uint64)
    acc = sets[0]
    for s in sets[1:]:
        if combine == "and":
            acc = np.intersect1d(acc, s)
        else:
            acc = np.union1d(acc, s)
    return acc.astype(np.uint64)
