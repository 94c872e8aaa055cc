// parquet_writer.h — minimal native Parquet writer for compaction output
// (DESIGN.md §10 / SURVEY §8(f) row 1). Writes exactly the reference writer's
// metric-SST layout (storage.rs:193-298 contract): 5 flat REQUIRED columns
// (series_id u64, timestamp i64, value f64, __seq__ u64, __reserved__ u64),
// row groups of `row_group` rows, one PLAIN uncompressed data page v1 per
// chunk, min/max statistics, thrift-compact footer. Readable by parquet-rs /
// pyarrow (validated in tests) and by our own reader.
#pragma once
#include <cstdint>
#include <string>

namespace hx {

// columns are caller-provided arrays of n rows; seq is constant per file
// (the compacted file's sequence — see hx_compact's closure precondition).
// Returns empty string on success, else an error message.
std::string write_metric_sst(const std::string& path, const uint64_t* series,
                             const int64_t* ts, const double* value,
                             uint64_t seq, int64_t n, int64_t row_group);

// General flat-table writer (same page/footer layout) for the auxiliary
// tables the RFC defines (metrics/series/tags/index,
// docs/rfcs/20240827-metric-engine.md:86-137). Fixed 8-byte columns
// (INT64/DOUBLE, optionally UINT_64-converted) and BYTE_ARRAY columns
// (PLAIN: u32 length + bytes; offsets[n+1] into `bytes`).
struct WriterCol {
    const char* name;
    int32_t physical;        // 2 INT64, 5 DOUBLE, 6 BYTE_ARRAY
    int32_t converted;       // -1 none, 14 UINT_64
    const void* data;        // 8-byte array, or bytes blob for BYTE_ARRAY
    const int64_t* offsets;  // BYTE_ARRAY only: n+1 offsets into data
};

std::string write_table_sst(const std::string& path, const WriterCol* cols,
                            int32_t n_cols, int64_t n, int64_t row_group);

// per-row variable-length seq variant of the metric writer (general
// compaction outputs, executor.rs:155-222 keep_builtin path)
std::string write_metric_sst_seqs(const std::string& path,
                                  const uint64_t* series, const int64_t* ts,
                                  const double* value, const uint64_t* seqs,
                                  int64_t n, int64_t row_group);

}  // namespace hx
