// parquet_writer.cpp — see parquet_writer.h. Thrift compact PROTOCOL writer
// (field ids per parquet-format/src/main/thrift/parquet.thrift; the inverse
// of thrift_compact.h's reader).
#include "parquet_writer.h"

#include <algorithm>
#include <cstdio>
#include <cstring>
#include <vector>

namespace hx {
namespace {

struct TW {  // thrift compact writer
    std::vector<uint8_t> buf;
    void byte(uint8_t b) { buf.push_back(b); }
    void varint(uint64_t v) {
        while (v >= 0x80) {
            byte(uint8_t(v) | 0x80);
            v >>= 7;
        }
        byte(uint8_t(v));
    }
    void zigzag(int64_t v) { varint((uint64_t(v) << 1) ^ uint64_t(v >> 63)); }
    // field header inside a struct (delta-encoded ids; caller tracks last)
    void field(int16_t& last, int16_t id, uint8_t type) {
        int16_t delta = id - last;
        if (delta > 0 && delta <= 15) {
            byte(uint8_t(delta << 4) | type);
        } else {
            byte(type);
            zigzag(id);
        }
        last = id;
    }
    void stop() { byte(0); }
    void i32(int16_t& last, int16_t id, int64_t v) {
        field(last, id, 5);
        zigzag(v);
    }
    void i64f(int16_t& last, int16_t id, int64_t v) {
        field(last, id, 6);
        zigzag(v);
    }
    void binary(int16_t& last, int16_t id, const void* p, size_t n) {
        field(last, id, 8);
        varint(n);
        const uint8_t* b = (const uint8_t*)p;
        buf.insert(buf.end(), b, b + n);
    }
    void str(int16_t& last, int16_t id, const std::string& s) {
        binary(last, id, s.data(), s.size());
    }
    void list_header(int16_t& last, int16_t id, uint8_t elem_type, size_t n) {
        field(last, id, 9);
        if (n < 15) {
            byte(uint8_t(n << 4) | elem_type);
        } else {
            byte(0xF0 | elem_type);
            varint(n);
        }
    }
};

struct ColSpec {
    const char* name;
    int32_t physical;    // 2 = INT64, 5 = DOUBLE
    int32_t converted;   // -1 none; 14 = UINT_64
};

const ColSpec kCols[5] = {
    {"series_id", 2, 14}, {"timestamp", 2, -1}, {"value", 5, -1},
    {"__seq__", 2, 14},   {"__reserved__", 2, 14},
};  // 14 = ConvertedType::UINT_64

void schema_element(TW& w, const ColSpec& c) {
    int16_t last = 0;
    w.i32(last, 1, c.physical);            // type
    w.i32(last, 3, 0);                     // repetition REQUIRED
    w.str(last, 4, c.name);                // name
    if (c.converted >= 0) w.i32(last, 6, c.converted);
    w.stop();
}

void statistics(TW& w, const std::string& mn, const std::string& mx) {
    int16_t last = 0;
    w.binary(last, 5, mx.data(), mx.size());  // max_value
    w.binary(last, 6, mn.data(), mn.size());  // min_value
    w.stop();
}

struct ChunkMeta {
    int64_t data_page_offset;
    int64_t total_size;
    int64_t num_values;
    std::string mn, mx;
    bool has_stats = true;
};

void column_chunk(TW& w, const char* name, int32_t physical,
                  const ChunkMeta& m) {
    int16_t last = 0;
    w.i64f(last, 2, m.data_page_offset);  // file_offset (deprecated, required)
    // meta_data struct (field 3)
    w.field(last, 3, 12);
    {
        int16_t l2 = 0;
        w.i32(l2, 1, physical);                        // type
        w.list_header(l2, 2, 5, 1);                    // encodings: [PLAIN]
        w.zigzag(0);
        w.list_header(l2, 3, 8, 1);                    // path_in_schema
        w.varint(strlen(name));
        w.buf.insert(w.buf.end(), (const uint8_t*)name,
                     (const uint8_t*)name + strlen(name));
        w.i32(l2, 4, 0);                               // codec UNCOMPRESSED
        w.i64f(l2, 5, m.num_values);
        w.i64f(l2, 6, m.total_size);                   // uncompressed
        w.i64f(l2, 7, m.total_size);                   // compressed
        w.i64f(l2, 9, m.data_page_offset);
        if (m.has_stats) {
            w.field(l2, 12, 12);                       // statistics
            statistics(w, m.mn, m.mx);
        }
        w.stop();
    }
    w.stop();
}

std::vector<uint8_t> page_header(int32_t n_values, int32_t payload) {
    TW w;
    int16_t last = 0;
    w.i32(last, 1, 0);        // type DATA_PAGE
    w.i32(last, 2, payload);  // uncompressed_page_size
    w.i32(last, 3, payload);  // compressed_page_size
    w.field(last, 5, 12);     // data_page_header
    {
        int16_t l2 = 0;
        w.i32(l2, 1, n_values);
        w.i32(l2, 2, 0);  // encoding PLAIN
        w.i32(l2, 3, 3);  // definition_level_encoding RLE
        w.i32(l2, 4, 3);  // repetition_level_encoding RLE
        w.stop();
    }
    w.stop();
    return std::move(w.buf);
}

template <typename T>
void minmax_bytes(const T* v, int64_t n, std::string& mn, std::string& mx) {
    T lo = v[0], hi = v[0];
    for (int64_t i = 1; i < n; i++) {
        if (v[i] < lo) lo = v[i];
        if (v[i] > hi) hi = v[i];
    }
    mn.assign((const char*)&lo, 8);
    mx.assign((const char*)&hi, 8);
}

}  // namespace

std::string write_metric_sst(const std::string& path, const uint64_t* series,
                             const int64_t* ts, const double* value,
                             uint64_t seq, int64_t n, int64_t row_group) {
    if (n <= 0) return "write_metric_sst: no rows";
    FILE* f = fopen(path.c_str(), "wb");
    if (!f) return "write_metric_sst: cannot open " + path;
    auto fail = [&](const char* m) {
        fclose(f);
        remove(path.c_str());
        return std::string(m);
    };
    if (fwrite("PAR1", 1, 4, f) != 4) return fail("write failed");
    int64_t off = 4;

    std::vector<uint64_t> seq_col;   // constant per file (closure precondition)
    std::vector<uint64_t> zeros;
    struct RG {
        int64_t num_rows;
        ChunkMeta cols[5];
    };
    std::vector<RG> rgs;
    for (int64_t base = 0; base < n; base += row_group) {
        int64_t rows = std::min<int64_t>(row_group, n - base);
        if ((int64_t)seq_col.size() < rows) {
            seq_col.assign(rows, seq);
            zeros.assign(rows, 0);
        }
        const void* data[5] = {series + base, ts + base, value + base,
                               seq_col.data(), zeros.data()};
        RG rg;
        rg.num_rows = rows;
        for (int c = 0; c < 5; c++) {
            int32_t payload = int32_t(rows * 8);
            auto hdr = page_header((int32_t)rows, payload);
            ChunkMeta& m = rg.cols[c];
            m.data_page_offset = off;
            m.num_values = rows;
            m.total_size = int64_t(hdr.size()) + payload;
            switch (c) {
                case 0: case 3: case 4:
                    minmax_bytes((const uint64_t*)data[c], rows, m.mn, m.mx);
                    break;
                case 1:
                    minmax_bytes((const int64_t*)data[c], rows, m.mn, m.mx);
                    break;
                case 2:
                    minmax_bytes((const double*)data[c], rows, m.mn, m.mx);
                    break;
            }
            m.has_stats = true;
            if (fwrite(hdr.data(), 1, hdr.size(), f) != hdr.size())
                return fail("write failed");
            if (fwrite(data[c], 8, rows, f) != size_t(rows))
                return fail("write failed");
            off += m.total_size;
        }
        rgs.push_back(rg);
    }

    // footer: FileMetaData
    TW w;
    int16_t last = 0;
    w.i32(last, 1, 1);  // version
    w.list_header(last, 2, 12, 6);  // schema: root + 5 leaves
    {
        int16_t l2 = 0;   // root group
        w.str(l2, 4, "schema");
        w.i32(l2, 5, 5);  // num_children
        w.stop();
    }
    for (const auto& c : kCols) schema_element(w, c);
    w.i64f(last, 3, n);  // num_rows
    w.list_header(last, 4, 12, rgs.size());
    for (const auto& rg : rgs) {
        int16_t l2 = 0;
        w.list_header(l2, 1, 12, 5);
        int64_t total = 0;
        for (int c = 0; c < 5; c++) total += rg.cols[c].total_size;
        for (int c = 0; c < 5; c++)
            column_chunk(w, kCols[c].name, kCols[c].physical, rg.cols[c]);
        w.i64f(l2, 2, total);        // total_byte_size
        w.i64f(l2, 3, rg.num_rows);  // num_rows
        w.stop();
    }
    w.str(last, 6, "horaedb-amd hx_compact writer");
    // column_orders: TYPE_ORDER for every leaf — required for readers to
    // trust min_value/max_value (parquet.thrift ColumnOrder union)
    w.list_header(last, 7, 12, 5);
    for (int c = 0; c < 5; c++) {
        int16_t l2 = 0;
        w.field(l2, 1, 12);  // TypeDefinedOrder (empty struct)
        w.stop();
        w.stop();
    }
    w.stop();

    uint32_t flen = (uint32_t)w.buf.size();
    if (fwrite(w.buf.data(), 1, flen, f) != flen) return fail("write failed");
    if (fwrite(&flen, 4, 1, f) != 1) return fail("write failed");
    if (fwrite("PAR1", 1, 4, f) != 4) return fail("write failed");
    if (fclose(f) != 0) return "write_metric_sst: close failed";
    return "";
}

// ---------------------------------------------------------------------------
// General flat-table writer (BYTE_ARRAY-capable) for the RFC's auxiliary
// tables (rfc:86-137) and per-row-seq compaction outputs. Same page/footer
// layout as write_metric_sst: one PLAIN uncompressed data page v1 per
// chunk, row groups of `row_group` rows, min/max statistics (omitted for
// byte values longer than 64 B), thrift-compact footer with column_orders.
// ---------------------------------------------------------------------------
std::string write_table_sst(const std::string& path, const WriterCol* cols,
                            int32_t n_cols, int64_t n, int64_t row_group) {
    if (n <= 0 || n_cols <= 0) return "write_table_sst: no rows/cols";
    FILE* f = fopen(path.c_str(), "wb");
    if (!f) return "write_table_sst: cannot open " + path;
    auto fail = [&](const char* m) {
        fclose(f);
        remove(path.c_str());
        return std::string(m);
    };
    for (int c = 0; c < n_cols; c++) {
        const WriterCol& wc = cols[c];
        if (wc.physical != 2 && wc.physical != 5 && wc.physical != 6)
            return fail("write_table_sst: unsupported physical type");
        if (wc.physical == 6 && !wc.offsets)
            return fail("write_table_sst: BYTE_ARRAY needs offsets");
    }
    if (fwrite("PAR1", 1, 4, f) != 4) return fail("write failed");
    int64_t off = 4;

    struct RG {
        int64_t num_rows;
        std::vector<ChunkMeta> cols;
    };
    std::vector<RG> rgs;
    std::vector<uint8_t> payload;
    for (int64_t base = 0; base < n; base += row_group) {
        int64_t rows = std::min<int64_t>(row_group, n - base);
        RG rg;
        rg.num_rows = rows;
        rg.cols.resize(n_cols);
        for (int c = 0; c < n_cols; c++) {
            const WriterCol& wc = cols[c];
            ChunkMeta& m = rg.cols[c];
            m.num_values = rows;
            m.data_page_offset = off;
            const uint8_t* body = nullptr;
            size_t body_len = 0;
            if (wc.physical == 6) {
                const uint8_t* bytes = (const uint8_t*)wc.data;
                payload.clear();
                m.has_stats = true;
                std::string mn, mx;
                for (int64_t r = base; r < base + rows; r++) {
                    int64_t b0 = wc.offsets[r], b1 = wc.offsets[r + 1];
                    if (b1 < b0) return fail("write_table_sst: bad offsets");
                    uint32_t len = (uint32_t)(b1 - b0);
                    payload.insert(payload.end(), (const uint8_t*)&len,
                                   (const uint8_t*)&len + 4);
                    payload.insert(payload.end(), bytes + b0, bytes + b1);
                    std::string v((const char*)bytes + b0, (size_t)len);
                    if (r == base || v < mn) mn = v;
                    if (r == base || v > mx) mx = v;
                }
                if (mn.size() > 64 || mx.size() > 64) m.has_stats = false;
                m.mn = mn;
                m.mx = mx;
                body = payload.data();
                body_len = payload.size();
            } else {
                body = (const uint8_t*)wc.data + base * 8;
                body_len = size_t(rows) * 8;
                if (wc.physical == 5)
                    minmax_bytes((const double*)wc.data + base, rows, m.mn,
                                 m.mx);
                else if (wc.converted == 14)
                    minmax_bytes((const uint64_t*)wc.data + base, rows, m.mn,
                                 m.mx);
                else
                    minmax_bytes((const int64_t*)wc.data + base, rows, m.mn,
                                 m.mx);
            }
            auto hdr = page_header((int32_t)rows, (int32_t)body_len);
            m.total_size = int64_t(hdr.size()) + int64_t(body_len);
            if (fwrite(hdr.data(), 1, hdr.size(), f) != hdr.size())
                return fail("write failed");
            if (body_len &&
                fwrite(body, 1, body_len, f) != body_len)
                return fail("write failed");
            off += m.total_size;
        }
        rgs.push_back(std::move(rg));
    }

    TW w;
    int16_t last = 0;
    w.i32(last, 1, 1);  // version
    w.list_header(last, 2, 12, size_t(n_cols) + 1);
    {
        int16_t l2 = 0;
        w.str(l2, 4, "schema");
        w.i32(l2, 5, n_cols);
        w.stop();
    }
    for (int c = 0; c < n_cols; c++) {
        const WriterCol& wc = cols[c];
        int16_t l2 = 0;
        w.i32(l2, 1, wc.physical);
        w.i32(l2, 3, 0);  // REQUIRED
        w.str(l2, 4, wc.name);
        if (wc.converted >= 0) w.i32(l2, 6, wc.converted);
        w.stop();
    }
    w.i64f(last, 3, n);
    w.list_header(last, 4, 12, rgs.size());
    for (const auto& rg : rgs) {
        int16_t l2 = 0;
        w.list_header(l2, 1, 12, n_cols);
        int64_t total = 0;
        for (int c = 0; c < n_cols; c++) total += rg.cols[c].total_size;
        for (int c = 0; c < n_cols; c++)
            column_chunk(w, cols[c].name, cols[c].physical, rg.cols[c]);
        w.i64f(l2, 2, total);
        w.i64f(l2, 3, rg.num_rows);
        w.stop();
    }
    w.str(last, 6, "horaedb-amd hx table writer");
    w.list_header(last, 7, 12, n_cols);
    for (int c = 0; c < n_cols; c++) {
        int16_t l2 = 0;
        w.field(l2, 1, 12);
        w.stop();
        w.stop();
    }
    w.stop();

    uint32_t flen = (uint32_t)w.buf.size();
    if (fwrite(w.buf.data(), 1, flen, f) != flen) return fail("write failed");
    if (fwrite(&flen, 4, 1, f) != 1) return fail("write failed");
    if (fwrite("PAR1", 1, 4, f) != 4) return fail("write failed");
    if (fclose(f) != 0) return "write_table_sst: close failed";
    return "";
}

std::string write_metric_sst_seqs(const std::string& path,
                                  const uint64_t* series, const int64_t* ts,
                                  const double* value, const uint64_t* seqs,
                                  int64_t n, int64_t row_group) {
    std::vector<uint64_t> zeros(n, 0);
    WriterCol cols[5] = {
        {"series_id", 2, 14, series, nullptr},
        {"timestamp", 2, -1, ts, nullptr},
        {"value", 5, -1, value, nullptr},
        {"__seq__", 2, 14, seqs, nullptr},
        {"__reserved__", 2, 14, zeros.data(), nullptr},
    };
    return write_table_sst(path, cols, 5, n, row_group);
}

}  // namespace hx
