/* horaedb_hx.h — C-ABI drop-in boundary for HoraeDB's metric-engine scan/
 * aggregate hot path, MI355X-native (gfx950 HIP kernels behind this ABI).
 *
 * Each entry point names the reference interface it replaces (apache/horaedb
 * `main`, paths relative to /root/reference/src). The reference surface kept
 * (SURVEY.md §8(b)):
 *   trait ColumnarStorage { schema/write/scan/compact }   columnar_storage/src/storage.rs:76-89
 *   ScanRequest { range, predicate, projections }         columnar_storage/src/storage.rs:65-70
 *   trait MergeOperator { merge(batch) -> batch }         columnar_storage/src/operator.rs:30-34
 *
 * Conventions: all structs are POD; caller owns all inputs; callee allocates
 * outputs, freed by the matching hx_*_free; status codes + hx_last_error()
 * (thread-local string). Concurrency: scan-side calls (hx_prepare /
 * hx_scan_agg / hx_scan) may run from many threads of one handle; a given
 * hx_prepared serves ONE call at a time; catalog mutations (hx_write,
 * hx_compact) require external serialization against all other calls, and
 * hx_find_ssts results are invalidated by the next hx_find_ssts on the
 * same handle. All compute is GPU-resident — if no
 * MI355X/HIP runtime is available every scan entry returns HX_ERR_NO_GPU
 * (there is no CPU fallback in this library).
 */
#ifndef HORAEDB_HX_H
#define HORAEDB_HX_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef int32_t hx_status;
enum {
    HX_OK             = 0,
    HX_ERR_IO         = 1,   /* file missing/unreadable */
    HX_ERR_FORMAT     = 2,   /* not a parquet SST / unsupported layout */
    HX_ERR_UNSUPPORTED= 3,   /* encoding/codec outside round-1 scope */
    HX_ERR_NO_GPU     = 4,   /* HIP runtime/device unavailable */
    HX_ERR_HIP        = 5,   /* HIP call failed (see hx_last_error) */
    HX_ERR_INVALID    = 6,   /* bad argument */
    HX_ERR_SCHEMA     = 7,   /* SST schema violates the metric contract */
};

/* Thread-local description of the last error from this thread. */
const char* hx_last_error(void);

/* ---- time range: [start, end), ms — types.rs:46-133 (TimeRange) -------- */
typedef struct { int64_t start; int64_t end; } hx_time_range;

/* ---- SST descriptor — sst.rs:154-160 (SstFile/FileMeta) ---------------- */
typedef struct {
    const char* path;       /* parquet file written by the reference writer */
    uint64_t    sequence;   /* __seq__ of every row in the file (= file id,
                               sst.rs:39-46); dedup winner = max sequence    */
} hx_sst_desc;

/* ---- predicates — ScanRequest.predicate (storage.rs:65-70) ------------- */
typedef enum {
    HX_PRED_SERIES_IN = 1,  /* series_id ∈ set (tag predicate materialized as
                               a series-id set, BASELINE config 3)           */
} hx_pred_kind;

typedef struct {
    int32_t kind;                   /* hx_pred_kind */
    const uint64_t* series_ids;     /* HX_PRED_SERIES_IN: sorted or not */
    size_t          n_series;
} hx_pred;

/* ---- scan spec — ScanRequest (storage.rs:65-70) ------------------------ */
typedef struct {
    hx_time_range   range;      /* ts-range predicate + SST pruning          */
    const hx_pred*  preds;      /* extra predicates, may be NULL             */
    size_t          n_preds;
    const hx_sst_desc* ssts;    /* explicit SST list; NULL => all SSTs the
                                   handle discovered under {store}/data that
                                   overlap `range` (Manifest::find_ssts,
                                   manifest/mod.rs:165-172)                  */
    size_t          n_ssts;
    const int32_t*  projection; /* hx_scan only: user-column indices, NULL =
                                   all user columns (builtins stripped —
                                   types.rs:203-239)                         */
    size_t          n_projection;
} hx_scan_spec;

/* ---- aggregate spec — no reference counterpart;
 * semantics pinned by BASELINE.json configs + rfc:218-231 ----------------- */
enum {
    HX_AGG_SUM   = 1u << 0,
    HX_AGG_COUNT = 1u << 1,
    HX_AGG_MIN   = 1u << 2,
    HX_AGG_MAX   = 1u << 3,
    HX_AGG_AVG   = 1u << 4,   /* finalized as sum/count */
};
typedef struct {
    uint32_t ops;         /* OR of HX_AGG_*  (over the value column) */
    int64_t  bucket_ms;   /* 0: group by series_id; >0: by (series_id,
                             timestamp/bucket_ms) — config 5 downsample */
} hx_agg_spec;

typedef struct {
    const int32_t* device_ids;  /* HIP device ordinals; NULL => {0} */
    int32_t        n_devices;
} hx_device_set;

/* ---- results ----------------------------------------------------------- */
/* Aggregate result: one row per group, sorted by (series_id, bucket) —
 * the reference stream's PK order (read.rs:479-494). Arrays are owned by the
 * table; NULL when the op was not requested. */
typedef struct hx_result_table {
    size_t          n_groups;
    const uint64_t* series_id;
    const int64_t*  bucket;     /* bucket index (ts/bucket_ms); NULL if bucket_ms==0 */
    const double*   sum;
    const uint64_t* count;
    const double*   vmin;
    const double*   vmax;
    const double*   avg;
} hx_result_table;

/* Streaming columnar batch for the non-aggregating parity mode — the
 * SendableRecordBatchStream analog (storage.rs:84, read.rs:349-385).
 * Columns follow the projection order; builtin __seq__/__reserved__ are
 * stripped as MergeStream does (read.rs:330-343). Column buffers are valid
 * only during the callback. */
/* variable-length byte column payload (Binary value schema): for a column
 * of type 3, the batch's cols[i] points at ONE hx_bytes_col; row r's value
 * is bytes[offsets[r] .. offsets[r+1]). */
typedef struct {
    const int64_t* offsets;       /* n_rows + 1 */
    const uint8_t* bytes;
} hx_bytes_col;

typedef struct {
    size_t   n_rows;
    size_t   n_cols;
    const void* const* cols;      /* col i: n_rows × 8B elements, or ONE
                                     hx_bytes_col when col_types[i] == 3    */
    const int32_t*     col_types; /* 0=u64, 1=i64(ts ms), 2=f64, 3=bytes    */
} hx_col_batch;
typedef int32_t (*hx_batch_cb)(void* ctx, const hx_col_batch* batch); /* nonzero => stop */

/* ---- lifecycle --------------------------------------------------------- */
typedef struct hx_handle   hx_handle;
typedef struct hx_prepared hx_prepared;

/* Replaces ObjectBasedStorage::try_new (storage.rs:138-187): opens a store
 * rooted at `store_path` ({store}/data/*.sst), reads every SST footer into
 * an in-memory catalog (FileMeta: rows, size, ts min/max — sst.rs:154-160).
 * segment_duration_ms partitions SSTs into time segments exactly as
 * storage.rs:106-136 does (server default 12h, server/config.rs:53).
 * Sequence of each file = numeric file stem ({id}.sst, sst.rs:193-205). */
hx_status hx_open(const char* store_path, int64_t segment_duration_ms,
                  hx_handle** out);
void      hx_close(hx_handle*);

/* ColumnarStorage::schema (storage.rs:76-89; StorageSchema contract
 * types.rs:150-240): the store's column layout — user columns (primary keys
 * first) plus the appended builtin __seq__/__reserved__ UInt64 columns. */
typedef struct {
    const char* name;
    int32_t     col_type;     /* 0=u64, 1=i64(ts ms), 2=f64 */
    int32_t     is_primary_key;
    int32_t     is_builtin;
} hx_col_desc;
hx_status hx_schema(hx_handle*, const hx_col_desc** out, size_t* n_out,
                    size_t* n_primary_keys);

/* Catalog introspection (Manifest::find_ssts, manifest/mod.rs:165-172). */
hx_status hx_find_ssts(hx_handle*, hx_time_range range,
                       const hx_sst_desc** out, size_t* n_out);

/* Stage the scan's column chunks into device HBM (file IO + footer/page
 * parse + row-group pruning + PCIe upload). Untimed prep: the timed hot
 * path starts at hx_exec_agg with inputs resident in HBM. */
hx_status hx_prepare(hx_handle*, const hx_scan_spec*, const hx_device_set*,
                     hx_prepared** out);
void      hx_prepared_free(hx_prepared*);

/* The hot path: decode + filter + dedup-merge + group-by aggregate, fully
 * on-GPU. Replaces scan (storage.rs:335-370) + read plan (read.rs:429-494)
 * + MergeExec (read.rs:100-391) + the aggregate the north star adds. */
hx_status hx_exec_agg(hx_prepared*, const hx_agg_spec*, hx_result_table** out);
void      hx_result_free(hx_result_table*);

/* One-shot convenience: prepare + exec_agg + release staging. */
hx_status hx_scan_agg(hx_handle*, const hx_scan_spec*, const hx_agg_spec*,
                      const hx_device_set*, hx_result_table** out);

/* Non-aggregating parity mode: the merged, deduplicated, PK-sorted row
 * stream itself (ColumnarStorage::scan semantics, storage.rs:335-370),
 * delivered as columnar batches through `cb`. GPU decode/filter/merge with
 * host readback; for parity testing, not the bench path. */
hx_status hx_scan(hx_handle*, const hx_scan_spec*, const hx_device_set*,
                  hx_batch_cb cb, void* ctx);

/* Compaction (ColumnarStorage::compact, storage.rs:76-89; executor
 * semantics executor.rs:155-222): GPU-merges the ts-overlap closure of the
 * SSTs overlapping `range` into ONE new SST (fresh file id = old max + 1),
 * updates the catalog, unlinks the inputs (add-before-delete). The closure
 * guarantees no remaining file shares primary keys with the inputs, so the
 * output carries one constant __seq__ without changing future merges. */
hx_status hx_compact(hx_handle*, hx_time_range range, const hx_device_set*,
                     uint64_t* out_new_seq);

/* General compaction of an EXPLICIT input set (the executor's Task{inputs},
 * compaction/mod.rs:26-36 + executor.rs:155-222 keep_builtin path): merges
 * and deduplicates ONLY among the named files and writes one SST whose rows
 * RETAIN their per-row __seq__ values, so later merges against files
 * outside the set still order correctly (MergeStream reads __seq__ from the
 * row, read.rs:289-343). No closure precondition. The scan path detects
 * such mixed-seq files from the __seq__ column statistics and compares
 * per-row sequences during cross-SST dedup. */
hx_status hx_compact_files(hx_handle*, const uint64_t* input_seqs,
                           size_t n_inputs, const hx_device_set*,
                           uint64_t* out_new_seq);

/* UpdateMode (config.rs:166-172): 0 = Overwrite (LastValueOperator), 1 =
 * Append (BytesMergeOperator — requires a Binary value column,
 * operator.rs:47-111). Append changes hx_scan's merge: equal-PK rows'
 * value bytes CONCATENATE in ascending __seq__ order. */
hx_status hx_set_update_mode(hx_handle*, int32_t mode);

/* ColumnarStorage::write (storage.rs:76-89, :307-333): stable PK sort of
 * the batch (sort_batch, storage.rs:244-256), file id allocation
 * (= sequence), one new SST via the native writer, catalog add. Ingest-side
 * prep (not the GPU scan hot path). enable_check enforces the
 * segment-crossing check (storage.rs:309-316). */
hx_status hx_write(hx_handle*, const uint64_t* series, const int64_t* ts,
                   const double* value, int64_t n_rows, int32_t enable_check,
                   uint64_t* out_seq);

/* Native SST writer (the compaction output path; PLAIN uncompressed,
 * row-group 8192 contract — storage.rs:193-298). Exposed for tests/ingest. */
hx_status hx_write_sst(const char* path, const uint64_t* series,
                       const int64_t* ts, const double* value, uint64_t seq,
                       int64_t n_rows, int64_t row_group);

/* ---- inverted index (RFC docs/rfcs/20240827-metric-engine.md:86-137) ----
 * The reference's planned index tables (its metric_engine index module is
 * an uncompiled skeleton; semantics pinned by the RFC): an `index` table
 * with rows (MetricID u64, TagKey bytes, TagValue bytes, TSID u64), stored
 * as Parquet SSTs under {store}/index/, rows sorted by
 * (tag_key, tag_value, tsid). A tag-equality predicate resolves to the
 * union of its postings across index SSTs; multiple predicates AND
 * (intersect) or OR (union). The resulting TSID set feeds the scan's
 * series-membership predicate (hx_scan_spec preds / hx_prepare series set),
 * closing the rfc "tag filter -> TSID -> data" query path on GPU. */
typedef struct {
    const char* tag_key;      /* NUL-terminated byte strings */
    const char* tag_value;
} hx_tag_pred;

/* Append one index SST (rows need not be pre-sorted; the writer sorts by
 * (tag_key, tag_value, tsid) — the RFC's index PK order). */
hx_status hx_index_write(hx_handle*, const uint64_t* metric_id,
                         const char* const* tag_keys,
                         const char* const* tag_values,
                         const uint64_t* tsids, int64_t n_rows,
                         uint64_t* out_seq);

/* Evaluate tag-equality predicates on `device`: GPU BYTE_ARRAY decode +
 * postings filter + set combine (AND = sorted intersection, OR = union).
 * Returns the sorted distinct TSID set (callee-allocated; free with
 * hx_tsids_free). */
hx_status hx_index_query(hx_handle*, const hx_tag_pred* preds, size_t n_preds,
                         int combine_and, int device, uint64_t** out_tsids,
                         size_t* n_out);
void hx_tsids_free(uint64_t* tsids);

/* ---- introspection for bench/tests ------------------------------------ */
typedef struct {
    double  exec_ms;          /* wall of last hx_exec_agg (HIP events)      */
    double  agg_kernel_ms;    /* k_scan_agg launch time (HIP events)        */
    double  decode_kernel_ms; /* delta/snappy decode kernels, if any        */
    int64_t rows_scanned;     /* rows examined (pre-filter)                 */
    int64_t rows_matched;     /* rows passing filter+dedup                  */
    int64_t bytes_staged;     /* page payload bytes resident in HBM         */
    double  stage_ms;         /* last hx_prepare wall (IO+parse+PCIe)       */
} hx_exec_stats;
hx_status hx_get_stats(hx_prepared*, hx_exec_stats* out);

#ifdef __cplusplus
}
#endif
#endif /* HORAEDB_HX_H */
