# oracle — TEST INFRASTRUCTURE ONLY.
#
# CPU restatement (numpy + pyarrow) of the reference scan/merge/aggregate
# semantics (apache/horaedb metric engine). This package is the
# parity checker for the MI355X product path (horaedb_amd + libhoraedb_hx.so);
# it is never shipped, never called by the product path, and never the thing
# measured — only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline
# leg may import it.
#
# Parity pinning: Parquet decode arithmetic lives in the
# unvendored third-party crates parquet-rs/arrow-rs 53.2.0 (Cargo.lock);
# in-repo the reference pins results only at the ColumnarStorage boundary.
# Decode here is delegated to pyarrow 25.0.0 (same format, Parquet spec);
# merge/dedup/schema semantics are restated from
#   read.rs:262-343 (MergeStream), operator.rs:37-111 (merge operators),
#   types.rs:150-240 (StorageSchema), storage.rs:189-298 (writer),
#   docs/rfcs/20240827-metric-engine.md:218-231 (data model)
# and pinned against the reference's own golden tests re-derived in
# tests/golden (see oracle/golden_gen.py).
from .scan import (  # noqa: F401
    MERGE_LAST,
    MERGE_APPEND,
    SstBatch,
    merge_scan,
    scan_agg,
    read_sst,
    truncate_by,
    fill_required_projections,
    scan_rows,
)
