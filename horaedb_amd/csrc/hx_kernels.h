// hx_kernels.h — host-side launcher API implemented in kernels.hip (keeps
// device code in one TU; no -fgpu-rdc needed).
#pragma once
#include <hip/hip_runtime.h>
#include "hx_device.h"

namespace hx {

struct CompactOut {
    uint64_t* series;
    long long* bucket;
    double* sum;
    unsigned long long* cnt;
    double* vmin;
    double* vmax;
    unsigned long long* n_out;
};

hipError_t launch_decode_delta(hipStream_t s, const uint8_t* blob, uint8_t* dec,
                               const DeltaPageDesc* pages, uint32_t n_pages,
                               unsigned long long* err_flag);
hipError_t launch_rledict(hipStream_t s, const uint8_t* blob, uint8_t* dec,
                          const RleDictPageDesc* pages, uint32_t n_pages,
                          unsigned long long* err_flag);
hipError_t launch_snappy(hipStream_t s, const uint8_t* blob, uint8_t* dec,
                         const SnappyPageDesc* pages, uint32_t n_pages,
                         unsigned long long* err_flag);
hipError_t launch_copy_u64(hipStream_t s, const uint8_t* blob, uint8_t* dec,
                           const CopyDesc* descs, uint32_t n_descs);
hipError_t launch_scan_agg(hipStream_t s, const AggParams& p, uint32_t grid);
struct GangParams;  // kernels.hip internal; caller provides param buffers
hipError_t launch_scan_agg_gang(hipStream_t s, const AggParams& p,
                                uint32_t gang_size, bool minmax,
                                GangParams* h_params, GangParams* d_params);
// Series-range mode: sample series for quantile boundaries,
// precompute per-(block,sst) row bounds, then the LDS-table range kernel.
hipError_t launch_sample_series(hipStream_t s, const RgDesc* rgs,
                                uint32_t n_rgs, const uint8_t* blob,
                                const uint8_t* dec, uint64_t* out);
hipError_t launch_range_bounds(hipStream_t s, const AggParams& p,
                               const RangeAux& r, uint64_t* out);
hipError_t launch_scan_agg_range2(hipStream_t s, const AggParams& p,
                                  const RangeAux& r, bool minmax);
hipError_t launch_scan_agg_range(hipStream_t s, const AggParams& p,
                                 const RangeAux& r, bool minmax);
hipError_t launch_scan_rows(hipStream_t s, const AggParams& p,
                            uint32_t rg_first, uint32_t rg_last,
                            uint64_t* out_series, long long* out_ts,
                            double* out_value, unsigned long long* cursor,
                            unsigned long long cap,
                            uint64_t* out_seq = nullptr,
                            int32_t seq_rowidx = 0);
hipError_t launch_gather_multi(hipStream_t s,
                               const unsigned long long* const* srcs,
                               uint32_t n_arrays, const uint32_t* perm,
                               unsigned long long* dst, uint32_t n);
// scratch: device buffer of 2*grid_for(n_slots,256) u32 (two-phase compact
// counts+bases); null routes to the single-pass kernel.
hipError_t launch_compact(hipStream_t s, const AggTable& t, uint32_t n_slots,
                          uint32_t ops, int32_t key_claim, int64_t bucket_ms,
                          int64_t lo_bucket, uint32_t n_buckets,
                          uint32_t bstride, const uint8_t* bstore,
                          const CompactOut& o, uint32_t* scratch);
hipError_t launch_gather_u64(hipStream_t s, const unsigned long long* in,
                             const uint32_t* perm, unsigned long long* out,
                             uint32_t n);
hipError_t launch_avg(hipStream_t s, const double* sum,
                      const unsigned long long* cnt, double* avg, uint32_t n);
hipError_t launch_iota(hipStream_t s, uint32_t* out, uint32_t n);
hipError_t launch_init_slab(hipStream_t s, uint8_t* slab, uint32_t n_slots,
                            uint32_t stride);
hipError_t launch_init_rep(hipStream_t s, uint8_t* rep, size_t total_slots,
                           uint32_t stride, bool mm);
hipError_t launch_init_state_slab(hipStream_t s, uint8_t* slab,
                                  uint32_t n_slots, uint32_t stride, bool mm);
hipError_t launch_seg_keys(hipStream_t s, const long long* ts, long long seg_ms,
                           unsigned long long* keys, uint32_t n);
hipError_t launch_xor_sign(hipStream_t s, unsigned long long* buf, uint32_t n);

// inverted-index query kernels (rfc:86-137)
hipError_t launch_ba_offsets(hipStream_t s, const uint8_t* blob,
                             const BaPageDesc* pages, uint32_t n_pages,
                             uint64_t* out, unsigned long long* err_flag);
hipError_t launch_tag_filter(hipStream_t s, const TagFilterParams& f);
hipError_t launch_copy_bytes(hipStream_t s, const uint8_t* blob,
                             const uint64_t* handles, const int64_t* dst_off,
                             uint8_t* out, uint32_t n);
hipError_t launch_tsid_intersect(hipStream_t s, const uint64_t* a,
                                 unsigned long long n_a, const uint64_t* b,
                                 unsigned long long n_b, uint64_t* out,
                                 unsigned long long* cursor);
hipError_t launch_unique_u64(hipStream_t s, const uint64_t* in,
                             unsigned long long n, uint64_t* out,
                             unsigned long long* cursor);
hipError_t sort_keys_u64(hipStream_t s, const uint64_t* keys_in,
                         uint64_t* keys_out, size_t n, void** d_temp,
                         size_t* temp_bytes);

// rocPRIM stable LSD radix sort: sorts values (u32 perm) by u64 keys.
// temp buffer managed internally on the stream (hipMallocAsync-free impl).
hipError_t sort_pairs_u64(hipStream_t s, const uint64_t* keys_in,
                          uint64_t* keys_out, const uint32_t* vals_in,
                          uint32_t* vals_out, size_t n, void** d_temp,
                          size_t* temp_bytes);

}  // namespace hx
