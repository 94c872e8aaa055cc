// hx_device.h — structures shared between the host engine and the gfx950
// kernels. All offsets are byte offsets; bit 63 of an
// offset selects the decode blob (dec) over the staged page blob (blob).
#pragma once
#include <cstdint>

namespace hx {

// aggregate op bits (must mirror HX_AGG_* in include/horaedb_hx.h)
enum : uint32_t { HXK_SUM = 1u, HXK_COUNT = 2u, HXK_MIN = 4u, HXK_MAX = 8u,
                  HXK_AVG = 16u };

constexpr uint64_t OFF_DEC = 1ull << 63;   // offset lives in decode blob
constexpr uint64_t OFF_MASK = OFF_DEC - 1;

// One (SST, row group) unit: the fused kernel's workgroup granule.
// Row groups of one SST are staged in ascending row order; next_rg gives the
// SAME SST's next staged row group (for the adjacent-dedup boundary row),
// -1 if none. Grid order of rgs is a speed-only choice (host interleaves by
// row ordinal across SSTs for L3 locality of the group table).
struct RgDesc {
    uint64_t series_off;
    uint64_t ts_off;
    uint64_t val_off;
    uint32_t n_rows;
    uint32_t sst_id;
    int64_t  row_base;     // staged-row index base within the SST (dense)
    int32_t  next_rg;      // index into rg array, -1 = last staged rg of sst
    int32_t  _pad;
};

// Per-SST device info. overlap==1 => this SST belongs to a ts-overlap
// cluster: rows must check higher-seq SSTs of the cluster for
// equal PKs via binary search over their dense (series, ts) arrays.
struct SstDev {
    int32_t  cluster;        // overlap-cluster id, -1 = none
    int32_t  rank;           // seq rank within cluster (ascending), dense
    uint64_t dense_series;   // byte offset (dec blob) of staged series array
    uint64_t dense_ts;       // byte offset (dec blob) of staged ts array
    int64_t  n_staged;       // staged rows (dense array length)
    // per-row sequences (keep_builtin compaction outputs,
    // executor.rs:155-222): a file whose __seq__ column is NOT constant
    // carries dense_seq (staged per-row seqs); 0 = constant (seq below).
    uint64_t seq;            // the file sequence (constant-seq files)
    uint64_t dense_seq;      // byte offset (dec blob) of per-row seqs, or 0
};

// Overlap cluster: member SSTs sorted by ascending seq; members[] indexes
// the SstDev array. Flattened: cluster c owns members[first..first+n).
struct ClusterDev {
    int32_t first;
    int32_t n;
};

// Aggregate hash table: open addressing, linear probe.
// state: 0 empty, 1 claim in flight, 2 ready (key words valid).
struct AggTable {
    uint64_t* series;
    int64_t*  bucket;        // null if bucket_ms == 0
    uint32_t* state;
    double*   sum;
    unsigned long long* cnt;
    unsigned long long* vmin;  // ordered-u64 mapped f64 (bit-exact min/max)
    unsigned long long* vmax;
    uint32_t  mask;            // slots-1 (power of two)
    // key-claim AoS mode: slot i = slab[i*stride .. ) = {key, sum, cnt
    // [, min, max]} — probe + update touch ONE cache line instead of three
    uint8_t*  slab;            // non-null => AoS mode (series-only grouping)
    uint32_t  stride;          // 32 (sum/cnt) or 64 (with min/max)
    // optional per-XCD accumulator replicas (HX_XCD_REP): keys live in the
    // shared slab; each XCD's blocks accumulate into their own replica to
    // avoid cross-L2 line ping-pong; compact merges the replicas.
    uint8_t*  rep;             // 8 replicas x slots x rep_stride, or null
    uint32_t  rep_stride;      // 16 (sum,cnt) or 32 (+min,max)
};

struct AggParams {
    const RgDesc* rgs;
    uint32_t n_rgs;
    const SstDev* ssts;
    const ClusterDev* clusters;
    const int32_t* cluster_members;
    const uint8_t* blob;
    const uint8_t* dec;
    int64_t ts_lo, ts_hi;          // [lo, hi)
    // optional series-membership set (open hash, EMPTY sentinel chosen by host)
    const uint64_t* sset;
    uint32_t sset_mask;
    uint64_t sset_empty;
    int32_t use_sset;
    int64_t bucket_ms;             // 0 = group by series only
    uint32_t ops;                  // HX_AGG_* mask
    int32_t key_claim;             // 1 = one-CAS key-claim mode (no state word)
    int32_t skip;                  // debug bisect: 1 skip table, 2 also skip dedup
    AggTable table;
    // direct-indexed bucket mode (bucket_ms > 0, key-claim safe, bucket
    // range known from the scan range): table keyed by series only; each
    // slot owns a dense [n_buckets] accumulator row in bstore.
    int64_t lo_bucket;
    uint32_t n_buckets;            // 0 => generic state-word path
    uint32_t bstride;              // 16 {sum,cnt} or 32 {+min,max}
    uint8_t* bstore;
    unsigned long long fill_limit; // early-abort when fill exceeds this
    unsigned long long* fill;      // claimed slots
    unsigned long long* overflow;  // !=0 => rerun with a larger table
    unsigned long long* matched;   // rows surviving filter+dedup
    unsigned long long* fallback;  // LDS-table misses routed to global RMWs
    int32_t poll;                  // 0 = no saturation poll in range kernel
    int32_t _pad2;
};

// Series-range partitioned aggregation (k_scan_agg_range): the
// series space is split at equal-sample quantile boundaries; block b owns
// series in [bounds[b], bounds[b+1]) across ALL SSTs, accumulating into an
// LDS table flushed once — global-table RMWs drop from one per 64-row-
// window run to one per (block, series). Row bounds per (block, sst) are
// precomputed by k_range_bounds (64-ary parallel lower_bound) and cached.
struct RangeAux {
    const uint64_t* bounds;      // n_blocks+1 ascending series boundaries
    const uint64_t* bound_rows;  // (n_blocks+1) x n_ssts packed (pos<<32|row)
    const int32_t* sst_rgs;      // rg-desc indices grouped per SST, row order
    const int32_t* sst_rg_off;   // per SST: offset into sst_rgs
    const int32_t* sst_rg_cnt;   // per SST: count
    uint32_t n_ssts;
    uint32_t n_blocks;
    uint32_t ne;                 // LDS hash slots (power of two)
    // 1 = interpolation slots: first probe at (s-lo)*ne/(hi-lo) within the
    // block's series range — ascending heads probe ascending slots, so the
    // wave's DS accesses land on consecutive banks (conflict-free) instead
    // of random ones, and collisions stay rare (series values are hashes,
    // ~uniform in value). 0 = mix64 hashing.
    uint32_t interp;
};

// DELTA_BINARY_PACKED decode unit: one page -> dense i64 at dst_off (dec).
struct DeltaPageDesc {
    uint64_t src_off;     // page payload (bit63: in dec blob, e.g. post-snappy)
    uint64_t dst_off;     // dec blob byte offset for n_values i64
    uint32_t n_values;
    uint32_t src_len;
};

// RLE_DICTIONARY decode unit: one data page + its dictionary page ->
// dense 8-byte values at dst_off (dec). Indices are the Parquet RLE/
// bit-packed hybrid; dictionary is PLAIN 8-byte values.
struct RleDictPageDesc {
    uint64_t dict_off;    // PLAIN dictionary payload (bit63: dec blob)
    uint64_t idx_off;     // data page payload (starts with bit-width byte)
    uint64_t dst_off;     // dec blob
    uint32_t dict_n;
    uint32_t idx_len;
    uint32_t n_values;
    uint32_t _pad;
};

// Snappy decompress unit: one page.
struct SnappyPageDesc {
    uint64_t src_off;     // blob
    uint64_t dst_off;     // dec blob
    uint32_t comp_len;
    uint32_t uncomp_len;
};

// Plain copy unit (materialize dense arrays for the overlap path).
struct CopyDesc {
    uint64_t src_off;
    uint64_t dst_off;
    uint32_t n_values;    // 8-byte values
    uint32_t _pad;
};

// ---- inverted-index query units (rfc:86-137 index table) ------------------
// One PLAIN BYTE_ARRAY page (u32 length + bytes per value): walked into a
// packed (offset << 20 | len) per row at out[first_row..first_row+n).
struct BaPageDesc {
    uint64_t src_off;     // page payload offset in the index blob
    uint64_t src_len;
    uint32_t n_values;
    uint32_t _pad;
    int64_t first_row;    // global row index of this page's first value
};

// Tag-equality filter over decoded (key, value) offset arrays + tsid column.
struct TagFilterParams {
    const uint8_t* blob;          // index page blob
    const uint64_t* key_offlen;   // packed per-row (off << 20 | len)
    const uint64_t* val_offlen;
    const uint64_t* tsid;         // dense per-row tsids
    int64_t n_rows;
    const uint8_t* pred_key;      // predicate bytes (device)
    uint32_t pred_key_len;
    uint32_t pred_val_len;
    const uint8_t* pred_val;
    uint64_t* out;                // matching tsids (unordered)
    unsigned long long* cursor;
    unsigned long long cap;
};

}  // namespace hx
