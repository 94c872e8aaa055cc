// parquet_meta.h — host-side Parquet footer / page-header parsing for the
// metric-engine SST layout. Replaces the reference's use of
// parquet-rs metadata decoding at the reader factory (read.rs:78-93) and the
// FileMeta catalog entries (sst.rs:154-160).
#pragma once
#include <cstdint>
#include <string>
#include <vector>

namespace hx {

// Parquet enums (format spec values)
enum PhysicalType : int32_t { PT_BOOLEAN = 0, PT_INT32 = 1, PT_INT64 = 2,
                              PT_INT96 = 3, PT_FLOAT = 4, PT_DOUBLE = 5,
                              PT_BYTE_ARRAY = 6, PT_FIXED = 7 };
enum Encoding : int32_t { ENC_PLAIN = 0, ENC_RLE = 3, ENC_BIT_PACKED = 4,
                          ENC_DELTA_BINARY_PACKED = 5, ENC_DELTA_LENGTH_BA = 6,
                          ENC_DELTA_BYTE_ARRAY = 7, ENC_RLE_DICTIONARY = 8,
                          ENC_PLAIN_DICTIONARY = 2, ENC_BYTE_STREAM_SPLIT = 9 };
enum Codec : int32_t { CODEC_UNCOMPRESSED = 0, CODEC_SNAPPY = 1, CODEC_GZIP = 2,
                       CODEC_LZO = 3, CODEC_BROTLI = 4, CODEC_LZ4 = 5,
                       CODEC_ZSTD = 6, CODEC_LZ4_RAW = 7 };

struct ColumnChunkMeta {
    int32_t physical_type = -1;
    int32_t codec = 0;
    int64_t num_values = 0;
    int64_t total_compressed_size = 0;
    int64_t total_uncompressed_size = 0;
    int64_t data_page_offset = -1;
    int64_t dictionary_page_offset = -1;   // -1 = none
    bool has_stats = false;
    std::string stat_min, stat_max;        // min_value/max_value (LE bytes)
    // chunk byte range in file:
    int64_t chunk_start() const {
        return dictionary_page_offset >= 0 && dictionary_page_offset < data_page_offset
                   ? dictionary_page_offset : data_page_offset;
    }
};

struct RowGroupMeta {
    int64_t num_rows = 0;
    std::vector<ColumnChunkMeta> columns;  // schema leaf order
};

struct SchemaColumn {
    std::string name;
    int32_t physical_type = -1;
    bool required = true;
};

struct FileMetadata {
    int64_t num_rows = 0;
    std::vector<SchemaColumn> columns;     // leaves, in order
    std::vector<RowGroupMeta> row_groups;
    std::string created_by;
};

// Parse a Parquet footer (whole-file buffer OR just the tail containing the
// footer; `file_size` = real file size, buf covers [file_size-len, file_size)).
// Throws std::runtime_error on malformed input.
FileMetadata parse_footer(const uint8_t* tail, size_t tail_len, int64_t file_size);

// One data/dictionary page found by walking a column chunk's page headers.
struct PageDesc {
    int32_t page_type;        // 0 data v1, 2 dictionary, 3 data v2
    int32_t encoding;
    int32_t num_values;
    int64_t payload_off;      // absolute file offset of page payload
    int32_t compressed_size;  // payload bytes in file
    int32_t uncompressed_size;
    int32_t def_level_bytes;  // levels preceding payload (v2; v1 OPTIONAL
                              // uncompressed: 4-byte-prefixed RLE block)
    int32_t is_compressed = 1; // data page v2 is_compressed flag (v1: always)
    int64_t null_count = -1;  // v1 statistics null_count (-1 = absent)
};

// Walk page headers of a column chunk (buf = the chunk's bytes; base_off =
// its absolute file offset). Stops after `num_values` data values seen.
// optional_col + codec let v1 pages of OPTIONAL columns account for their
// embedded definition-level block (nulls themselves are rejected).
std::vector<PageDesc> walk_pages(const uint8_t* buf, size_t len,
                                 int64_t base_off, int64_t num_values,
                                 bool optional_col = false,
                                 int32_t codec = 0);

// little-endian i64 from an 8-byte statistics value
int64_t stat_i64(const std::string& s);

}  // namespace hx
