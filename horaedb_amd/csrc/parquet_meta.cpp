// parquet_meta.cpp — see parquet_meta.h. Thrift struct field ids follow the
// Apache Parquet format IDL (parquet-format/src/main/thrift/parquet.thrift).
#include "parquet_meta.h"
#include "thrift_compact.h"

#include <cstring>

namespace hx {

int64_t stat_i64(const std::string& s) {
    if (s.size() != 8) throw std::runtime_error("stat value not 8 bytes");
    int64_t v;
    std::memcpy(&v, s.data(), 8);
    return v;  // little-endian host
}

namespace {

struct SchemaElem {
    std::string name;
    int32_t type = -1;
    int32_t repetition = -1;  // 0 required, 1 optional, 2 repeated
    int32_t num_children = 0;
};

SchemaElem parse_schema_element(ThriftReader& r) {
    SchemaElem e;
    int16_t last = 0, id; uint8_t t;
    while (r.field(last, id, t)) {
        switch (id) {
            case 1: e.type = int32_t(r.zigzag()); break;        // Type
            case 3: e.repetition = int32_t(r.zigzag()); break;  // repetition_type
            case 4: e.name = r.binary(); break;
            case 5: e.num_children = int32_t(r.zigzag()); break;
            default: r.skip(t);
        }
    }
    return e;
}

void parse_statistics(ThriftReader& r, ColumnChunkMeta& c) {
    int16_t last = 0, id; uint8_t t;
    std::string depr_max, depr_min;
    bool has_new_max = false, has_new_min = false;
    while (r.field(last, id, t)) {
        switch (id) {
            case 1: depr_max = r.binary(); break;       // max (deprecated)
            case 2: depr_min = r.binary(); break;       // min (deprecated)
            case 5: c.stat_max = r.binary(); has_new_max = true; break;
            case 6: c.stat_min = r.binary(); has_new_min = true; break;
            default: r.skip(t);
        }
    }
    if (!has_new_max && !depr_max.empty()) c.stat_max = depr_max;
    if (!has_new_min && !depr_min.empty()) c.stat_min = depr_min;
    c.has_stats = !c.stat_min.empty() && !c.stat_max.empty();
}

ColumnChunkMeta parse_column_meta(ThriftReader& r) {
    ColumnChunkMeta c;
    int16_t last = 0, id; uint8_t t;
    while (r.field(last, id, t)) {
        switch (id) {
            case 1: c.physical_type = int32_t(r.zigzag()); break;
            case 4: c.codec = int32_t(r.zigzag()); break;
            case 5: c.num_values = r.zigzag(); break;
            case 6: c.total_uncompressed_size = r.zigzag(); break;
            case 7: c.total_compressed_size = r.zigzag(); break;
            case 9: c.data_page_offset = r.zigzag(); break;
            case 11: c.dictionary_page_offset = r.zigzag(); break;
            case 12: parse_statistics(r, c); break;
            default: r.skip(t);
        }
    }
    return c;
}

ColumnChunkMeta parse_column_chunk(ThriftReader& r) {
    ColumnChunkMeta c;
    int16_t last = 0, id; uint8_t t;
    while (r.field(last, id, t)) {
        switch (id) {
            case 3: c = parse_column_meta(r); break;  // meta_data struct
            default: r.skip(t);
        }
    }
    return c;
}

RowGroupMeta parse_row_group(ThriftReader& r) {
    RowGroupMeta rg;
    int16_t last = 0, id; uint8_t t;
    while (r.field(last, id, t)) {
        switch (id) {
            case 1: {  // columns: list<ColumnChunk>
                uint8_t et; uint32_t n;
                r.list_header(et, n);
                for (uint32_t i = 0; i < n; i++)
                    rg.columns.push_back(parse_column_chunk(r));
                break;
            }
            case 3: rg.num_rows = r.zigzag(); break;
            default: r.skip(t);
        }
    }
    return rg;
}

}  // namespace

FileMetadata parse_footer(const uint8_t* tail, size_t tail_len, int64_t file_size) {
    if (tail_len < 8) throw std::runtime_error("file too small for parquet footer");
    const uint8_t* magic = tail + tail_len - 4;
    if (std::memcmp(magic, "PAR1", 4) != 0)
        throw std::runtime_error("missing PAR1 magic (not a parquet SST)");
    uint32_t flen;
    std::memcpy(&flen, tail + tail_len - 8, 4);
    if (flen + 8 > tail_len)
        throw std::runtime_error("footer longer than provided tail");
    const uint8_t* fbuf = tail + tail_len - 8 - flen;
    (void)file_size;

    ThriftReader r(fbuf, flen);
    FileMetadata m;
    int16_t last = 0, id; uint8_t t;
    while (r.field(last, id, t)) {
        switch (id) {
            case 2: {  // schema: list<SchemaElement>, depth-first; leaves only
                uint8_t et; uint32_t n;
                r.list_header(et, n);
                for (uint32_t i = 0; i < n; i++) {
                    SchemaElem e = parse_schema_element(r);
                    if (i == 0) continue;  // root group
                    if (e.num_children > 0)
                        throw std::runtime_error("nested schema unsupported (metric SSTs are flat)");
                    m.columns.push_back({e.name, e.type, e.repetition == 0});
                }
                break;
            }
            case 3: m.num_rows = r.zigzag(); break;
            case 4: {  // row_groups
                uint8_t et; uint32_t n;
                r.list_header(et, n);
                for (uint32_t i = 0; i < n; i++)
                    m.row_groups.push_back(parse_row_group(r));
                break;
            }
            case 6: m.created_by = r.binary(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::vector<PageDesc> walk_pages(const uint8_t* buf, size_t len,
                                 int64_t base_off, int64_t num_values,
                                 bool optional_col, int32_t codec) {
    std::vector<PageDesc> pages;
    size_t pos = 0;
    int64_t seen = 0;
    while (seen < num_values) {
        if (pos >= len) throw std::runtime_error("page walk ran past chunk");
        ThriftReader r(buf + pos, len - pos);
        PageDesc pd{};
        pd.page_type = -1;
        pd.is_compressed = 1;
        int32_t uncomp = 0, comp = 0;
        int16_t last = 0, id; uint8_t t;
        while (r.field(last, id, t)) {
            switch (id) {
                case 1: pd.page_type = int32_t(r.zigzag()); break;
                case 2: uncomp = int32_t(r.zigzag()); break;
                case 3: comp = int32_t(r.zigzag()); break;
                case 5: {  // DataPageHeader (v1)
                    int16_t l2 = 0, id2; uint8_t t2;
                    while (r.field(l2, id2, t2)) {
                        switch (id2) {
                            case 1: pd.num_values = int32_t(r.zigzag()); break;
                            case 2: pd.encoding = int32_t(r.zigzag()); break;
                            case 5: {  // Statistics
                                int16_t l3 = 0, id3; uint8_t t3;
                                while (r.field(l3, id3, t3)) {
                                    if (id3 == 3)
                                        pd.null_count = r.zigzag();
                                    else
                                        r.skip(t3);
                                }
                                break;
                            }
                            default: r.skip(t2);
                        }
                    }
                    break;
                }
                case 7: {  // DictionaryPageHeader
                    int16_t l2 = 0, id2; uint8_t t2;
                    while (r.field(l2, id2, t2)) {
                        switch (id2) {
                            case 1: pd.num_values = int32_t(r.zigzag()); break;
                            case 2: pd.encoding = int32_t(r.zigzag()); break;
                            default: r.skip(t2);
                        }
                    }
                    break;
                }
                case 8: {  // DataPageHeaderV2
                    int16_t l2 = 0, id2; uint8_t t2;
                    while (r.field(l2, id2, t2)) {
                        switch (id2) {
                            case 1: pd.num_values = int32_t(r.zigzag()); break;
                            case 4: pd.encoding = int32_t(r.zigzag()); break;
                            case 5: pd.def_level_bytes = int32_t(r.zigzag()); break;
                            case 6: pd.def_level_bytes += int32_t(r.zigzag()); break;
                            case 7: pd.is_compressed = (t2 == 1); break;
                            default: r.skip(t2);
                        }
                    }
                    break;
                }
                default: r.skip(t);
            }
        }
        if (comp < 0 || uncomp < 0)
            throw std::runtime_error("negative page size");
        size_t hdr_len = r.offset(buf + pos);
        // every page's payload (including the last) must fit the chunk
        if (pos + hdr_len > len || size_t(comp) > len - pos - hdr_len)
            throw std::runtime_error("page payload exceeds chunk bounds");
        // v2 def/rep level bytes precede the payload and must fit it
        if (pd.def_level_bytes < 0 || pd.def_level_bytes > comp ||
            pd.def_level_bytes > uncomp)
            throw std::runtime_error("level bytes exceed page size");
        pd.payload_off = base_off + int64_t(pos + hdr_len);
        pd.compressed_size = comp;
        pd.uncompressed_size = uncomp;
        if (pd.page_type == 0 && optional_col) {
            // v1 data page of an OPTIONAL column: definition levels are a
            // 4-byte-length-prefixed RLE block at the START of the payload
            // (inside the compressed stream when the page is compressed —
            // that case is not staged here). The metric schema's rows have
            // no nulls in practice (arrow-rs writes def levels for nullable
            // columns anyway); reject real nulls loudly.
            if (pd.null_count > 0)
                throw std::runtime_error(
                    "v1 data page has nulls (null_count > 0): nullable "
                    "values are outside the metric scan contract");
            if (codec != CODEC_UNCOMPRESSED)
                throw std::runtime_error(
                    "v1 OPTIONAL column with compressed pages: def levels "
                    "are inside the compressed stream (unsupported; write "
                    "REQUIRED columns or uncompressed pages)");
            if (size_t(comp) < 4)
                throw std::runtime_error("v1 def-level prefix truncated");
            const uint8_t* pl = buf + pos + hdr_len;
            uint32_t rle_len = (uint32_t)pl[0] | ((uint32_t)pl[1] << 8) |
                               ((uint32_t)pl[2] << 16) |
                               ((uint32_t)pl[3] << 24);
            if (4 + (int64_t)rle_len >= comp)
                throw std::runtime_error("v1 def-level block exceeds page");
            pd.def_level_bytes = int32_t(4 + rle_len);
        }
        pos += hdr_len + size_t(comp);
        if (pd.page_type == 0 || pd.page_type == 3) seen += pd.num_values;
        pages.push_back(pd);
    }
    return pages;
}

}  // namespace hx
