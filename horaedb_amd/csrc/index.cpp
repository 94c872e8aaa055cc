// index.cpp — inverted-index tables and the GPU tag→TSID query path
// (C-ABI: hx_index_write / hx_index_query in include/horaedb_hx.h).
//
// Restates the reference RFC's index design
// (docs/rfcs/20240827-metric-engine.md:86-137): an `index` table
// (MetricID u64, TagKey bytes, TagValue bytes, TSID u64) managed as Parquet
// SSTs (the RFC: "采用 Table 来管理上述结构，不同字段可以直接对应 parquet
// 的一个列" — each field is a Parquet column), rows PK-sorted by
// (tag_key, tag_value, tsid). The reference's own index module is an
// uncompiled skeleton (metric_engine/src/index/mod.rs — SURVEY §2), so the
// query semantics are pinned by the RFC text: a label filter resolves to
// the TSID postings of its (TagKey, TagValue), then feeds the data scan.
//
// GPU path (HBM-bound byte work, no MFMA): page payloads staged to HBM,
// k_ba_offsets walks the PLAIN BYTE_ARRAY length prefixes (pages in
// parallel), k_tag_filter does the per-row byte-equality postings filter
// with wave-ballot compaction, rocPRIM radix sort + k_unique_u64 build each
// predicate's sorted distinct TSID set, and k_tsid_intersect combines AND
// predicates. OR is the sorted union. Index SSTs are written uncompressed
// PLAIN (our writer); compressed index pages are rejected loudly.
#include "../../include/horaedb_hx.h"
#include "hx_internal.h"
#include "parquet_meta.h"
#include "parquet_writer.h"
#include "hx_device.h"
#include "hx_kernels.h"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstring>
#include <dirent.h>
#include <fcntl.h>
#include <numeric>
#include <string>
#include <sys/stat.h>
#include <unistd.h>
#include <vector>

using hx_int::set_error;
using hx_int::store_path;

#define IHIP_TRY(expr)                                                        \
    do {                                                                      \
        hipError_t _e = (expr);                                               \
        if (_e != hipSuccess)                                                 \
            return set_error(HX_ERR_HIP, std::string(#expr) + ": " +          \
                                             hipGetErrorString(_e));          \
    } while (0)

namespace {

std::string index_dir(hx_handle* h) { return store_path(h) + "/index"; }

// list {store}/index/{seq}.sst ascending by seq
std::vector<std::pair<uint64_t, std::string>> list_index_ssts(hx_handle* h) {
    std::vector<std::pair<uint64_t, std::string>> out;
    std::string dir = index_dir(h);
    DIR* d = opendir(dir.c_str());
    if (!d) return out;
    while (dirent* e = readdir(d)) {
        std::string name = e->d_name;
        if (name.size() < 5 || name.substr(name.size() - 4) != ".sst")
            continue;
        char* endp = nullptr;
        uint64_t seq = strtoull(name.c_str(), &endp, 10);
        if (!endp || std::string(endp) != ".sst") continue;
        out.emplace_back(seq, dir + "/" + name);
    }
    closedir(d);
    std::sort(out.begin(), out.end());
    return out;
}

struct DevBuf {
    void* p = nullptr;
    ~DevBuf() {
        if (p) (void)hipFree(p);
    }
    hipError_t alloc(size_t n) { return hipMalloc(&p, n ? n : 1); }
};

}  // namespace

extern "C" hx_status hx_index_write(hx_handle* h, const uint64_t* metric_id,
                                    const char* const* tag_keys,
                                    const char* const* tag_values,
                                    const uint64_t* tsids, int64_t n,
                                    uint64_t* out_seq) {
    if (!h || !tag_keys || !tag_values || !tsids || n <= 0 || !out_seq)
        return set_error(HX_ERR_INVALID, "hx_index_write: bad argument");
    *out_seq = 0;
    // PK sort (tag_key, tag_value, tsid) — the RFC index table order
    std::vector<uint32_t> order(n);
    std::iota(order.begin(), order.end(), 0u);
    std::sort(order.begin(), order.end(), [&](uint32_t a, uint32_t b) {
        int c = strcmp(tag_keys[a], tag_keys[b]);
        if (c) return c < 0;
        c = strcmp(tag_values[a], tag_values[b]);
        if (c) return c < 0;
        return tsids[a] < tsids[b];
    });
    std::vector<uint64_t> mid(n), tid(n);
    std::vector<int64_t> koff(n + 1), voff(n + 1);
    std::string kbytes, vbytes;
    koff[0] = voff[0] = 0;
    for (int64_t i = 0; i < n; i++) {
        uint32_t r = order[i];
        mid[i] = metric_id ? metric_id[r] : 0;
        tid[i] = tsids[r];
        kbytes += tag_keys[r];
        vbytes += tag_values[r];
        koff[i + 1] = (int64_t)kbytes.size();
        voff[i + 1] = (int64_t)vbytes.size();
    }
    std::string dir = index_dir(h);
    mkdir(dir.c_str(), 0755);
    uint64_t seq = 1;
    for (auto& [sq, _] : list_index_ssts(h)) seq = std::max(seq, sq + 1);
    hx::WriterCol cols[4] = {
        {"metric_id", 2, 14, mid.data(), nullptr},
        {"tag_key", 6, -1, kbytes.data(), koff.data()},
        {"tag_value", 6, -1, vbytes.data(), voff.data()},
        {"tsid", 2, 14, tid.data(), nullptr},
    };
    std::string err = hx::write_table_sst(
        dir + "/" + std::to_string(seq) + ".sst", cols, 4, n, 8192);
    if (!err.empty()) return set_error(HX_ERR_IO, err);
    *out_seq = seq;
    return HX_OK;
}

extern "C" void hx_tsids_free(uint64_t* t) { free(t); }

extern "C" hx_status hx_index_query(hx_handle* h, const hx_tag_pred* preds,
                                    size_t n_preds, int combine_and,
                                    int device, uint64_t** out_tsids,
                                    size_t* n_out) {
    if (!h || !preds || n_preds == 0 || !out_tsids || !n_out)
        return set_error(HX_ERR_INVALID, "hx_index_query: bad argument");
    *out_tsids = nullptr;
    *n_out = 0;
    auto files = list_index_ssts(h);
    if (files.empty())
        return set_error(HX_ERR_INVALID,
                         "hx_index_query: no index tables under " +
                             index_dir(h));
    IHIP_TRY(hipSetDevice(device));
    hipStream_t s = nullptr;  // default stream: queries are one-shot

    // ---- catalog + staging layout: for every (file, rg) overlapping any
    // pred key by tag_key stats, stage the tag_key/tag_value page payloads
    // and the PLAIN tsid values into one host blob -------------------------
    struct RgUnit {
        int64_t n_rows;
        uint64_t key_off, key_len;     // blob offsets (page payloads)
        uint64_t val_off, val_len;
        uint64_t tsid_off;             // blob offset of dense u64 payload
        int64_t first_row;
    };
    std::vector<RgUnit> units;
    std::vector<uint8_t> blob;
    int64_t total_rows = 0;
    auto align64 = [](size_t x) { return (x + 63) & ~size_t(63); };
    for (auto& [seq, path] : files) {
        int fd = open(path.c_str(), O_RDONLY);
        if (fd < 0) return set_error(HX_ERR_IO, "open " + path);
        off_t fsz = lseek(fd, 0, SEEK_END);
        uint8_t t8[8];
        if (fsz < 12 || pread(fd, t8, 8, fsz - 8) != 8) {
            close(fd);
            return set_error(HX_ERR_FORMAT, path + ": too small");
        }
        uint32_t flen;
        std::memcpy(&flen, t8, 4);
        if ((int64_t)flen + 8 > fsz) {
            close(fd);
            return set_error(HX_ERR_FORMAT, path + ": bad footer");
        }
        std::vector<uint8_t> tail(flen + 8);
        if (pread(fd, tail.data(), flen + 8, fsz - 8 - flen) !=
            (ssize_t)(flen + 8)) {
            close(fd);
            return set_error(HX_ERR_IO, path + ": footer read");
        }
        hx::FileMetadata m;
        try {
            m = hx::parse_footer(tail.data(), tail.size(), fsz);
        } catch (const std::exception& e) {
            close(fd);
            return set_error(HX_ERR_FORMAT, path + ": " + e.what());
        }
        int ck = -1, cv = -1, ct = -1;
        for (size_t i = 0; i < m.columns.size(); i++) {
            if (m.columns[i].name == "tag_key") ck = (int)i;
            if (m.columns[i].name == "tag_value") cv = (int)i;
            if (m.columns[i].name == "tsid") ct = (int)i;
        }
        if (ck < 0 || cv < 0 || ct < 0) {
            close(fd);
            return set_error(HX_ERR_SCHEMA,
                             path + ": index schema columns missing "
                                    "(rfc:86-137 index table)");
        }
        for (auto& rg : m.row_groups) {
            // prune by tag_key min/max statistics vs the predicate keys
            const auto& kcc = rg.columns[ck];
            if (kcc.has_stats && !kcc.stat_min.empty() &&
                !kcc.stat_max.empty()) {
                bool any = false;
                for (size_t p = 0; p < n_preds; p++) {
                    std::string k = preds[p].tag_key;
                    if (k >= kcc.stat_min && k <= kcc.stat_max) {
                        any = true;
                        break;
                    }
                }
                if (!any) continue;
            }
            RgUnit u{};
            u.n_rows = rg.num_rows;
            u.first_row = total_rows;
            for (int which = 0; which < 3; which++) {
                const auto& cc =
                    rg.columns[which == 0 ? ck : which == 1 ? cv : ct];
                if (cc.codec != hx::CODEC_UNCOMPRESSED) {
                    close(fd);
                    return set_error(HX_ERR_UNSUPPORTED,
                                     path + ": compressed index pages are "
                                            "not supported (write index "
                                            "tables uncompressed)");
                }
                std::vector<uint8_t> chunk(cc.total_compressed_size);
                if (pread(fd, chunk.data(), chunk.size(), cc.chunk_start()) !=
                    (ssize_t)chunk.size()) {
                    close(fd);
                    return set_error(HX_ERR_IO, path + ": chunk read");
                }
                const int col_i = which == 0 ? ck : which == 1 ? cv : ct;
                std::vector<hx::PageDesc> pages;
                try {
                    pages = hx::walk_pages(chunk.data(), chunk.size(),
                                           cc.chunk_start(), cc.num_values,
                                           !m.columns[col_i].required,
                                           cc.codec);
                } catch (const std::exception& e) {
                    close(fd);
                    return set_error(HX_ERR_FORMAT, path + ": " + e.what());
                }
                const hx::PageDesc* dp = nullptr;
                int nd = 0;
                for (auto& p : pages)
                    if (p.page_type == 0 || p.page_type == 3) {
                        dp = &p;
                        nd++;
                    }
                if (nd != 1 || !dp || dp->num_values != rg.num_rows) {
                    close(fd);
                    return set_error(HX_ERR_FORMAT,
                                     path + ": expected one data page per "
                                            "index chunk");
                }
                size_t in_chunk = size_t(dp->payload_off - cc.chunk_start()) +
                                  size_t(dp->def_level_bytes);
                size_t plen = size_t(dp->compressed_size) -
                              size_t(dp->def_level_bytes);
                if (in_chunk + plen > chunk.size()) {
                    close(fd);
                    return set_error(HX_ERR_FORMAT, path + ": page bounds");
                }
                size_t dst = align64(blob.size());
                blob.resize(dst + plen);
                std::memcpy(blob.data() + dst, chunk.data() + in_chunk, plen);
                if (which == 0) {
                    u.key_off = dst;
                    u.key_len = plen;
                } else if (which == 1) {
                    u.val_off = dst;
                    u.val_len = plen;
                } else {
                    if (plen != size_t(rg.num_rows) * 8) {
                        close(fd);
                        return set_error(HX_ERR_FORMAT,
                                         path + ": tsid payload size");
                    }
                    u.tsid_off = dst;
                }
            }
            total_rows += rg.num_rows;
            units.push_back(u);
        }
        close(fd);
    }
    if (total_rows == 0) {
        *out_tsids = (uint64_t*)malloc(1);
        *n_out = 0;
        return HX_OK;
    }

    // ---- device staging -------------------------------------------------
    DevBuf d_blob, d_pages, d_offk, d_offv, d_tsid_rows, d_scratch;
    IHIP_TRY(d_blob.alloc(blob.size()));
    IHIP_TRY(hipMemcpy(d_blob.p, blob.data(), blob.size(),
                       hipMemcpyHostToDevice));
    std::vector<hx::BaPageDesc> pdescs;
    for (auto& u : units) {
        pdescs.push_back({u.key_off, u.key_len, (uint32_t)u.n_rows, 0,
                          u.first_row});
        pdescs.push_back({u.val_off, u.val_len, (uint32_t)u.n_rows, 0,
                          u.first_row});
    }
    IHIP_TRY(d_pages.alloc(pdescs.size() * sizeof(hx::BaPageDesc)));
    IHIP_TRY(hipMemcpy(d_pages.p, pdescs.data(),
                       pdescs.size() * sizeof(hx::BaPageDesc),
                       hipMemcpyHostToDevice));
    IHIP_TRY(d_offk.alloc(size_t(total_rows) * 8));
    IHIP_TRY(d_offv.alloc(size_t(total_rows) * 8));
    // dense tsid rows: gather the per-unit payload slices into one array
    IHIP_TRY(d_tsid_rows.alloc(size_t(total_rows) * 8));
    for (auto& u : units)
        IHIP_TRY(hipMemcpyAsync(
            (uint8_t*)d_tsid_rows.p + size_t(u.first_row) * 8,
            (uint8_t*)d_blob.p + u.tsid_off, size_t(u.n_rows) * 8,
            hipMemcpyDeviceToDevice, s));
    // counters + error flag
    unsigned long long* d_ctr = nullptr;
    IHIP_TRY(hipMalloc((void**)&d_ctr, 4 * 8));
    DevBuf ctr_guard;
    ctr_guard.p = d_ctr;
    IHIP_TRY(hipMemsetAsync(d_ctr, 0, 32, s));
    // BYTE_ARRAY offsets: key pages are even indices, value pages odd —
    // both in one launch writing to separate arrays? The kernel writes one
    // array; run twice (key pages then value pages).
    std::vector<hx::BaPageDesc> kp, vp;
    for (size_t i = 0; i < pdescs.size(); i += 2) kp.push_back(pdescs[i]);
    for (size_t i = 1; i < pdescs.size(); i += 2) vp.push_back(pdescs[i]);
    DevBuf d_kp, d_vp;
    IHIP_TRY(d_kp.alloc(kp.size() * sizeof(hx::BaPageDesc)));
    IHIP_TRY(d_vp.alloc(vp.size() * sizeof(hx::BaPageDesc)));
    IHIP_TRY(hipMemcpy(d_kp.p, kp.data(), kp.size() * sizeof(hx::BaPageDesc),
                       hipMemcpyHostToDevice));
    IHIP_TRY(hipMemcpy(d_vp.p, vp.data(), vp.size() * sizeof(hx::BaPageDesc),
                       hipMemcpyHostToDevice));
    IHIP_TRY(hx::launch_ba_offsets(s, (const uint8_t*)d_blob.p,
                                   (const hx::BaPageDesc*)d_kp.p,
                                   (uint32_t)kp.size(), (uint64_t*)d_offk.p,
                                   d_ctr + 3));
    IHIP_TRY(hx::launch_ba_offsets(s, (const uint8_t*)d_blob.p,
                                   (const hx::BaPageDesc*)d_vp.p,
                                   (uint32_t)vp.size(), (uint64_t*)d_offv.p,
                                   d_ctr + 3));
    IHIP_TRY(hipStreamSynchronize(s));
    unsigned long long ba_err = 0;
    IHIP_TRY(hipMemcpy(&ba_err, d_ctr + 3, 8, hipMemcpyDeviceToHost));
    if (ba_err)
        return set_error(HX_ERR_FORMAT,
                         "index BYTE_ARRAY page walk failed (malformed "
                         "length prefix)");

    // ---- per-predicate postings -> sorted distinct TSIDs -----------------
    // scratch: match buffer + sort ping-pong (3 x total_rows u64)
    IHIP_TRY(d_scratch.alloc(size_t(total_rows) * 8 * 3));
    uint64_t* d_match = (uint64_t*)d_scratch.p;
    uint64_t* d_sorted = d_match + total_rows;
    uint64_t* d_uniq = d_sorted + total_rows;
    void* d_temp = nullptr;
    size_t temp_bytes = 0;
    struct TempGuard {
        void** p;
        ~TempGuard() {
            if (*p) (void)hipFree(*p);
        }
    } tg{&d_temp};

    DevBuf d_predbytes;
    size_t predcap = 0;
    for (size_t p = 0; p < n_preds; p++)
        predcap += strlen(preds[p].tag_key) + strlen(preds[p].tag_value);
    IHIP_TRY(d_predbytes.alloc(predcap + 1));
    std::string predblob;
    std::vector<std::pair<size_t, size_t>> pred_off;  // (key off, val off)
    for (size_t p = 0; p < n_preds; p++) {
        pred_off.emplace_back(predblob.size(),
                              predblob.size() + strlen(preds[p].tag_key));
        predblob += preds[p].tag_key;
        predblob += preds[p].tag_value;
    }
    IHIP_TRY(hipMemcpy(d_predbytes.p, predblob.data(), predblob.size() + 1,
                       hipMemcpyHostToDevice));

    DevBuf d_acc;                  // device running set (sorted)
    unsigned long long n_acc = 0;
    for (size_t p = 0; p < n_preds; p++) {
        IHIP_TRY(hipMemsetAsync(d_ctr, 0, 16, s));
        hx::TagFilterParams F{};
        F.blob = (const uint8_t*)d_blob.p;
        F.key_offlen = (const uint64_t*)d_offk.p;
        F.val_offlen = (const uint64_t*)d_offv.p;
        F.tsid = (const uint64_t*)d_tsid_rows.p;
        F.n_rows = total_rows;
        F.pred_key = (const uint8_t*)d_predbytes.p + pred_off[p].first;
        F.pred_key_len = (uint32_t)strlen(preds[p].tag_key);
        F.pred_val = (const uint8_t*)d_predbytes.p + pred_off[p].second;
        F.pred_val_len = (uint32_t)strlen(preds[p].tag_value);
        F.out = d_match;
        F.cursor = d_ctr;
        F.cap = (unsigned long long)total_rows;
        IHIP_TRY(hx::launch_tag_filter(s, F));
        IHIP_TRY(hipStreamSynchronize(s));
        unsigned long long n_match = 0;
        IHIP_TRY(hipMemcpy(&n_match, d_ctr, 8, hipMemcpyDeviceToHost));
        if (n_match > (unsigned long long)total_rows)
            n_match = (unsigned long long)total_rows;
        unsigned long long n_cur = 0;
        if (n_match) {
            IHIP_TRY(hx::sort_keys_u64(s, d_match, d_sorted, n_match,
                                       &d_temp, &temp_bytes));
            IHIP_TRY(hipMemsetAsync(d_ctr + 1, 0, 8, s));
            IHIP_TRY(hx::launch_unique_u64(s, d_sorted, n_match, d_uniq,
                                           d_ctr + 1));
            IHIP_TRY(hipStreamSynchronize(s));
            IHIP_TRY(hipMemcpy(&n_cur, d_ctr + 1, 8, hipMemcpyDeviceToHost));
            // unique output is unordered across blocks: sort it back
            IHIP_TRY(hx::sort_keys_u64(s, d_uniq, d_sorted, n_cur, &d_temp,
                                       &temp_bytes));
        }
        if (p == 0) {
            // 2x: the OR path concatenates before sort+unique
            IHIP_TRY(d_acc.alloc(size_t(total_rows) * 16));
            if (n_cur)
                IHIP_TRY(hipMemcpyAsync(d_acc.p, d_sorted, n_cur * 8,
                                        hipMemcpyDeviceToDevice, s));
            n_acc = n_cur;
        } else if (combine_and) {
            // intersect acc with this pred's set
            IHIP_TRY(hipMemsetAsync(d_ctr + 2, 0, 8, s));
            if (n_acc && n_cur) {
                IHIP_TRY(hx::launch_tsid_intersect(
                    s, (const uint64_t*)d_acc.p, n_acc, d_sorted, n_cur,
                    d_uniq, d_ctr + 2));
                IHIP_TRY(hipStreamSynchronize(s));
                unsigned long long n_i = 0;
                IHIP_TRY(hipMemcpy(&n_i, d_ctr + 2, 8,
                                   hipMemcpyDeviceToHost));
                IHIP_TRY(hx::sort_keys_u64(s, d_uniq, (uint64_t*)d_acc.p,
                                           n_i, &d_temp, &temp_bytes));
                n_acc = n_i;
            } else {
                n_acc = 0;
            }
        } else {
            // OR: merge-union via concat + sort + unique
            if (n_cur) {
                IHIP_TRY(hipMemcpyAsync((uint64_t*)d_acc.p + n_acc, d_sorted,
                                        n_cur * 8, hipMemcpyDeviceToDevice,
                                        s));
                n_acc += n_cur;
                IHIP_TRY(hx::sort_keys_u64(s, (const uint64_t*)d_acc.p,
                                           d_sorted, n_acc, &d_temp,
                                           &temp_bytes));
                IHIP_TRY(hipMemsetAsync(d_ctr + 1, 0, 8, s));
                IHIP_TRY(hx::launch_unique_u64(s, d_sorted, n_acc, d_uniq,
                                               d_ctr + 1));
                IHIP_TRY(hipStreamSynchronize(s));
                unsigned long long n_u = 0;
                IHIP_TRY(hipMemcpy(&n_u, d_ctr + 1, 8,
                                   hipMemcpyDeviceToHost));
                IHIP_TRY(hx::sort_keys_u64(s, d_uniq, (uint64_t*)d_acc.p,
                                           n_u, &d_temp, &temp_bytes));
                n_acc = n_u;
            }
        }
        if (combine_and && n_acc == 0 && p + 1 < n_preds) break;
    }
    IHIP_TRY(hipStreamSynchronize(s));
    uint64_t* host = (uint64_t*)malloc(std::max<size_t>(1, n_acc * 8));
    if (!host) return set_error(HX_ERR_IO, "hx_index_query: oom");
    if (n_acc) {
        hipError_t ce = hipMemcpy(host, d_acc.p, n_acc * 8,
                                  hipMemcpyDeviceToHost);
        if (ce != hipSuccess) {
            free(host);
            return set_error(HX_ERR_HIP, hipGetErrorString(ce));
        }
    }
    *out_tsids = host;
    *n_out = (size_t)n_acc;
    return HX_OK;
}
