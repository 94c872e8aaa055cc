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

}  // namespace hx
