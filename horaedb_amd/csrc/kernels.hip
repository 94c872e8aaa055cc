// kernels.hip — gfx950 (MI355X, CDNA4) kernels for the scan/aggregate hot
// path. HBM-bandwidth-bound: wave = 64, coalesced 8B loads,
// one workgroup per 8192-row row group; group table updated with device
// atomics (distinct addresses; L2/L3-resident by grid ordering).
// Replaces the decode/filter/merge arithmetic the reference runs in
// parquet-rs/arrow-rs/datafusion (SURVEY §2 third-party row) plus the
// north-star aggregate (no reference counterpart).
#include <hip/hip_runtime.h>
#include "hx_device.h"

namespace hx {

#define RLX __ATOMIC_RELAXED
#define AGT __HIP_MEMORY_SCOPE_AGENT

// sentinel for the one-CAS key-claim table mode (host proves, from column
// statistics, that no staged series_id equals it before enabling the mode)
#define KEY_EMPTY 0xFFFFFFFFFFFFFFFFull

__device__ __forceinline__ const uint8_t* hx_ptr(const uint8_t* blob,
                                                 const uint8_t* dec,
                                                 uint64_t off) {
    return ((off & OFF_DEC) ? dec : blob) + (off & OFF_MASK);
}

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

// monotone f64 -> u64 map: preserves total order, so u64 atomicMin/Max give
// bit-exact f64 min/max.
__device__ __forceinline__ unsigned long long f64_ordered(double v) {
    unsigned long long u = __double_as_longlong(v);
    return (u >> 63) ? ~u : (u | 0x8000000000000000ull);
}
__device__ __forceinline__ double ordered_f64(unsigned long long u) {
    return __longlong_as_double((u >> 63) ? (u & 0x7FFFFFFFFFFFFFFFull) : ~u);
}

__device__ __forceinline__ int64_t floordiv(int64_t a, int64_t b) {
    int64_t q = a / b;
    if ((a % b) != 0 && ((a < 0) != (b < 0))) q--;
    return q;
}

__device__ __forceinline__ bool sset_has(const AggParams& P, uint64_t s) {
    if (s == P.sset_empty) return false;  // host guarantees sentinel ∉ set
    uint32_t i = (uint32_t)mix64(s) & P.sset_mask;
    for (uint32_t probes = 0; probes <= P.sset_mask; ++probes) {
        uint64_t k = P.sset[i];
        if (k == s) return true;
        if (k == P.sset_empty) return false;
        i = (i + 1) & P.sset_mask;
    }
    return false;
}

// Cross-SST dedup: is row (s,t) of SST `me` (row sequence
// my_seq) shadowed by an equal PK with a HIGHER sequence in another SST of
// the same ts-overlap cluster? SSTs are PK-sorted (storage.rs:244-256), so
// a binary search per candidate member. Constant-seq members below my_seq
// are pruned outright; mixed-seq members (keep_builtin compaction outputs)
// compare the matched row's own __seq__ (read.rs:289-343 orders by
// (pk..., __seq__) read from the row).
__device__ bool shadowed(const AggParams& P, const SstDev& me,
                         uint64_t my_seq, uint64_t s, int64_t t) {
    ClusterDev c = P.clusters[me.cluster];
    for (int32_t j = c.first; j < c.first + c.n; ++j) {
        const SstDev o = P.ssts[P.cluster_members[j]];
        if (o.rank == me.rank) continue;
        if (!o.dense_seq && o.seq <= my_seq) continue;
        const uint64_t* S = (const uint64_t*)(P.dec + (o.dense_series & OFF_MASK));
        const int64_t* T = (const int64_t*)(P.dec + (o.dense_ts & OFF_MASK));
        int64_t lo = 0, hi = o.n_staged;
        while (lo < hi) {
            int64_t mid = (lo + hi) >> 1;
            uint64_t sm = S[mid];
            if (sm < s || (sm == s && T[mid] < t)) lo = mid + 1;
            else hi = mid;
        }
        if (lo < o.n_staged && S[lo] == s && T[lo] == t) {
            uint64_t o_seq = o.dense_seq
                ? ((const uint64_t*)(P.dec + (o.dense_seq & OFF_MASK)))[lo]
                : o.seq;
            if (o_seq > my_seq) return true;
        }
    }
    return false;
}

// row sequence of staged row (rg.row_base + r) of `sst`
__device__ __forceinline__ uint64_t row_seq(const AggParams& P,
                                            const SstDev& sst,
                                            const RgDesc& rg, uint32_t r) {
    return sst.dense_seq
        ? ((const uint64_t*)(P.dec + (sst.dense_seq & OFF_MASK)))
              [rg.row_base + r]
        : sst.seq;
}

// Open-addressing claim + update. state: 0 empty / 1 claiming / 2 ready.
// Claim via CAS on the state word (works for any 8/16-byte key — no key
// sentinel needed); claimer stores key words then releases state=2; readers
// are gated by the control dependency on state==2 (all table words accessed
// with agent-scope atomics => L2-coherent, no L1 staleness).
// accumulate into slot i: the shared slab, or this XCD's replica when
// per-XCD replication is on (the replica's lines stay exclusive to one L2
// — no cross-XCD ping-pong on the hot accumulator lines)
__device__ __forceinline__ void keycas_add(const AggParams& P, uint32_t i,
                                           int64_t b, double vsum,
                                           unsigned long long cnt, double mn,
                                           double mx) {
    uint8_t* slot;
    uint32_t off8, off16, off24, off32;
    if (P.n_buckets) {  // direct-indexed bucket row of this series slot
        // bounds check: lo_bucket/n_buckets derive from footer ts stats; an
        // SST whose stats under-report the data range must not index out of
        // the dense row (host sees overflow and retries the hashed path)
        const uint64_t bi = (uint64_t)(b - P.lo_bucket);
        if (bi >= (uint64_t)P.n_buckets) {
            __hip_atomic_fetch_add(P.overflow, 1ull, RLX, AGT);
            return;
        }
        slot = P.bstore + ((size_t)i * P.n_buckets + (size_t)bi) * P.bstride;
        off8 = 0; off16 = 8; off24 = 16; off32 = 24;
    } else if (P.table.rep) {
        uint32_t xcc;
        asm("s_getreg_b32 %0, hwreg(HW_REG_XCC_ID)" : "=s"(xcc));
        xcc &= 7u;
        slot = P.table.rep +
               ((size_t)xcc * (P.table.mask + 1ull) + i) * P.table.rep_stride;
        off8 = 0; off16 = 8; off24 = 16; off32 = 24;
    } else {
        slot = P.table.slab + (size_t)i * P.table.stride;
        off8 = 8; off16 = 16; off24 = 24; off32 = 32;
    }
    if (P.ops & (HXK_SUM | HXK_AVG)) atomicAdd((double*)(slot + off8), vsum);
    if (P.ops & (HXK_COUNT | HXK_AVG))
        atomicAdd((unsigned long long*)(slot + off16), cnt);
    if (P.ops & HXK_MIN)
        atomicMin((unsigned long long*)(slot + off24), f64_ordered(mn));
    if (P.ops & HXK_MAX)
        atomicMax((unsigned long long*)(slot + off32), f64_ordered(mx));
}

// Fast claim path (AoS slab): slot i = {key, sum, cnt[, min, max]} in one
// cache line. key claimed by one CAS against KEY_EMPTY (host proves via
// column statistics that no series == KEY_EMPTY; series-only grouping);
// probe is a plain cached load (a slot transitions KEY_EMPTY -> key exactly
// once per exec, so a stale read can only be KEY_EMPTY, which the CAS
// corrects).
__device__ __forceinline__ void agg_update_keycas(const AggParams& P, uint64_t s,
                                                  int64_t b, double vsum,
                                                  unsigned long long cnt,
                                                  double mn, double mx,
                                                  uint32_t hint_i = 0xFFFFFFFFu,
                                                  uint64_t hint_k = 0,
                                                  bool count_fill = true) {
    const uint32_t stride = P.table.stride;
    uint32_t i = (uint32_t)mix64(s) & P.table.mask;
    if (hint_i == i && hint_k == s) {  // prefetched probe already matched
        keycas_add(P, i, b, vsum, cnt, mn, mx);
        return;
    }
    const uint32_t probe_cap = P.table.mask < 4096u ? P.table.mask : 4096u;
    for (uint32_t probes = 0; probes <= probe_cap; ++probes) {
        uint8_t* slot = P.table.slab + (size_t)i * stride;
        uint64_t k = *(uint64_t*)slot;
        if (k == KEY_EMPTY) {
            uint64_t expected = KEY_EMPTY;
            if (__hip_atomic_compare_exchange_strong((uint64_t*)slot,
                    &expected, s, RLX, RLX, AGT)) {
                // P.fill is ONE word: per-claim increments serialize at the
                // coherence point and backpressure the whole vmcnt pipeline
                // (measured ~17 ms of the 24 ms launch). Callers that count
                // claims in bulk (the range kernels' flush) pass
                // count_fill = false and add per-block totals instead.
                if (count_fill)
                    __hip_atomic_fetch_add(P.fill, 1ull, RLX, AGT);
                k = s;
            } else {
                k = expected;
            }
        }
        if (k == s) {
            keycas_add(P, i, b, vsum, cnt, mn, mx);
            return;
        }
        i = (i + 1) & P.table.mask;
    }
    __hip_atomic_fetch_add(P.overflow, 1ull, RLX, AGT);
}

extern "C" __global__ void __launch_bounds__(256)
k_init_state_slab(uint8_t* slab, uint32_t n_slots, uint32_t stride,
                  uint32_t mm) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n_slots;
         i += blockDim.x * gridDim.x) {
        uint8_t* slot = slab + (size_t)i * stride;
        ((uint64_t*)slot)[0] = 0;   // state + pad
        ((uint64_t*)slot)[3] = 0;   // sum
        ((uint64_t*)slot)[4] = 0;   // cnt
        if (mm) {
            ((uint64_t*)slot)[5] = ~0ull;
            ((uint64_t*)slot)[6] = 0;
        }
    }
}

extern "C" __global__ void __launch_bounds__(256)
k_init_rep(uint8_t* rep, size_t total_slots, uint32_t stride, uint32_t mm) {
    for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total_slots;
         i += (size_t)blockDim.x * gridDim.x) {
        uint8_t* slot = rep + i * stride;
        ((uint64_t*)slot)[0] = 0;   // sum
        ((uint64_t*)slot)[1] = 0;   // cnt
        if (mm) {
            ((uint64_t*)slot)[2] = ~0ull;
            ((uint64_t*)slot)[3] = 0;
        }
    }
}

// slab init: key=KEY_EMPTY, sum=0, cnt=0, min=~0 (ordered), max=0
extern "C" __global__ void __launch_bounds__(256)
k_init_slab(uint8_t* slab, uint32_t n_slots, uint32_t stride) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n_slots;
         i += blockDim.x * gridDim.x) {
        uint8_t* slot = slab + (size_t)i * stride;
        ((uint64_t*)slot)[0] = KEY_EMPTY;
        ((uint64_t*)slot)[1] = 0;   // sum
        ((uint64_t*)slot)[2] = 0;   // cnt
        if (stride >= 64) {
            ((uint64_t*)slot)[3] = ~0ull;  // min (ordered)
            ((uint64_t*)slot)[4] = 0;      // max (ordered)
        }
    }
}

// General path (16-byte keys, e.g. (series,bucket)): claim via a CAS on a
// state word (0 empty / 1 claiming / 2 ready), then key words; readers are
// gated by the control dependency on state==2 (agent-scope atomics =>
// L2-bypassing, coherent at the memory side).
__device__ __forceinline__ void agg_update(const AggParams& P, uint64_t s,
                                           int64_t b, double vsum,
                                           unsigned long long cnt,
                                           double mn, double mx,
                                           uint32_t hint_i = 0xFFFFFFFFu,
                                           uint64_t hint_k = 0,
                                           bool count_fill = true) {
    if (P.key_claim) {
        agg_update_keycas(P, s, b, vsum, cnt, mn, mx, hint_i, hint_k,
                          count_fill);
        return;
    }
    // AoS slot: {state u32, pad u32, series u64, bucket i64, sum, cnt
    // [, min, max]} — probe + claim + update on one/two cache lines
    const uint32_t stride = P.table.stride;
    uint64_t h = mix64(s ^ ((uint64_t)b * 0xD1B54A32D192ED03ull));
    uint32_t i = (uint32_t)h & P.table.mask;
    const uint32_t probe_cap = P.table.mask < 4096u ? P.table.mask : 4096u;
    for (uint32_t probes = 0; probes <= probe_cap; ++probes) {
        uint8_t* slot = P.table.slab + (size_t)i * stride;
        uint32_t* stp = (uint32_t*)slot;
        uint32_t st = __hip_atomic_load(stp, RLX, AGT);
        if (st == 0) {
            uint32_t expected = 0;
            if (__hip_atomic_compare_exchange_strong(stp, &expected, 1u, RLX,
                                                     RLX, AGT)) {
                __hip_atomic_store((uint64_t*)(slot + 8), s, RLX, AGT);
                __hip_atomic_store((unsigned long long*)(slot + 16),
                                   (unsigned long long)b, RLX, AGT);
                __hip_atomic_fetch_add(P.fill, 1ull, RLX, AGT);
                __hip_atomic_store(stp, 2u, __ATOMIC_RELEASE, AGT);
                st = 2;
            } else {
                st = expected;
            }
        }
        if (st == 1) {  // claimer is storing the key; bounded spin
            uint32_t spins = 0;
            do {
                st = __hip_atomic_load(stp, RLX, AGT);
                if (++spins > (1u << 22)) {
                    __hip_atomic_fetch_add(P.overflow, 1ull, RLX, AGT);
                    return;
                }
            } while (st != 2);
        }
        if (__hip_atomic_load((uint64_t*)(slot + 8), RLX, AGT) == s &&
            (int64_t)__hip_atomic_load((unsigned long long*)(slot + 16), RLX,
                                       AGT) == b) {
            if (P.ops & (HXK_SUM | HXK_AVG))
                atomicAdd((double*)(slot + 24), vsum);
            if (P.ops & (HXK_COUNT | HXK_AVG))
                atomicAdd((unsigned long long*)(slot + 32), cnt);
            if (P.ops & HXK_MIN)
                atomicMin((unsigned long long*)(slot + 40), f64_ordered(mn));
            if (P.ops & HXK_MAX)
                atomicMax((unsigned long long*)(slot + 48), f64_ordered(mx));
            return;
        }
        i = (i + 1) & P.table.mask;
    }
    __hip_atomic_fetch_add(P.overflow, 1ull, RLX, AGT);
}

// Per-row filter + dedup (shared by both aggregate kernels).
// Returns alive; s/t are the row's PK. DESIGN.md §5.
__device__ __forceinline__ bool row_alive(const AggParams& P, const RgDesc& rg,
                                          const SstDev& sst, const uint64_t* S,
                                          const int64_t* T, uint32_t r,
                                          uint32_t n, uint64_t s, int64_t t) {
    bool alive = (t >= P.ts_lo) & (t < P.ts_hi);
    if (alive && P.use_sset) alive = sset_has(P, s);
    if (alive && P.skip < 2) {   // skip>=2: Append keeps every row
        // within-SST dedup: the LAST row of an equal-PK run survives
        // (LastValueOperator, operator.rs:37-44; plan order read.rs:456-480)
        bool dup = false;
        if (r + 1 < n) {
            dup = (S[r + 1] == s) & (T[r + 1] == t);
        } else if (rg.next_rg >= 0) {
            const RgDesc nx = P.rgs[rg.next_rg];
            uint64_t s2 = *(const uint64_t*)hx_ptr(P.blob, P.dec, nx.series_off);
            int64_t t2 = *(const int64_t*)hx_ptr(P.blob, P.dec, nx.ts_off);
            dup = (s2 == s) & (t2 == t);
        }
        if (!dup && sst.cluster >= 0)
            dup = shadowed(P, sst, row_seq(P, sst, rg, r), s, t);
        alive = !dup;
    }
    return alive;
}

// ---------------------------------------------------------------------------
// The headline fused kernel: decode(PLAIN in place) + ts-range/series-set
// filter + MergeExec dedup + hash group-by aggregate. One workgroup per row
// group, grid-stride; each thread strides rows (coalesced 8B column loads).
// ---------------------------------------------------------------------------
// Per-window worker for k_scan_agg: 64 consecutive rows on one wave —
// loads, filter, dedup, wave-segmented pre-reduction. Returns whether this
// lane must issue a table update (run head with survivors) via out params.
struct WinResult {
    uint64_t s;
    int64_t b;
    double vv, mn, mx;
    unsigned long long c;
    bool head;
    uint32_t hint_i;    // early-probed slot index (key-claim tables)
    uint64_t hint_k;    // its key value at probe time
};

__device__ __forceinline__ void scan_window(const AggParams& P,
                                            const RgDesc& rg,
                                            const SstDev& sst,
                                            const uint64_t* S,
                                            const int64_t* T, const double* V,
                                            uint32_t base, uint32_t n,
                                            int lane,
                                            unsigned long long& my_matched,
                                            WinResult& W) {
    const uint32_t r = base + threadIdx.x;
    const bool inb = r < n;
    const bool has_next = r + 1 < n;
    uint64_t s = KEY_EMPTY;
    int64_t t = 0;
    double v = 0.0;
    uint64_t s1 = 0;
    int64_t t1 = 0;
    bool alive = false;
    // issue ALL of the window's loads together (row, value, successor):
    // one memory round-trip per window instead of a 3-deep dependent chain
    if (inb) {
        t = T[r];
        s = S[r];
        v = V[r];
        if (has_next) {
            s1 = S[r + 1];
            t1 = T[r + 1];
        }
    }
    if (inb) {
        alive = (t >= P.ts_lo) & (t < P.ts_hi);
        if (alive && P.use_sset) alive = sset_has(P, s);
        if (alive && P.skip < 2) {
            bool dup = false;
            if (has_next) {
                dup = (s1 == s) & (t1 == t);
            } else if (rg.next_rg >= 0) {
                const RgDesc nx = P.rgs[rg.next_rg];
                uint64_t s2 = *(const uint64_t*)hx_ptr(P.blob, P.dec,
                                                       nx.series_off);
                int64_t t2 = *(const int64_t*)hx_ptr(P.blob, P.dec, nx.ts_off);
                dup = (s2 == s) & (t2 == t);
            }
            if (!dup && sst.cluster >= 0)
            dup = shadowed(P, sst, row_seq(P, sst, rg, r), s, t);
            alive = !dup;
        }
        if (!alive) v = 0.0;
    }
    int64_t b = (inb && P.bucket_ms) ? floordiv(t, P.bucket_ms) : 0;
    unsigned long long c = alive ? 1ull : 0ull;
    double vv = alive ? v : 0.0;
    double mn = alive ? v : HUGE_VAL;
    double mx = alive ? v : -HUGE_VAL;
    my_matched += c;
    // issue the table probe NOW: its L3 latency hides under the cross-lane
    // reduction, and a hit skips the dependent probe load in agg_update
    W.hint_i = 0xFFFFFFFFu;
    W.hint_k = 0;
    if (P.key_claim && alive) {
        W.hint_i = (uint32_t)mix64(s) & P.table.mask;
        W.hint_k = *(const uint64_t*)(P.table.slab +
                                      (size_t)W.hint_i * P.table.stride);
    }
    const uint64_t sp = __shfl_up(s, 1, 64);
    const int64_t bp = __shfl_up((long long)b, 1, 64);
    const bool head = (lane == 0) || sp != s || bp != b;
    bool done = false;
    for (int d = 1; d < 64; d++) {
        const uint64_t s2 = __shfl_down(s, d, 64);
        const long long b2 = __shfl_down((long long)b, d, 64);
        const double v2 = __shfl_down(vv, d, 64);
        const unsigned long long c2 = __shfl_down(c, d, 64);
        const double mn2 = __shfl_down(mn, d, 64);
        const double mx2 = __shfl_down(mx, d, 64);
        done = done || (lane + d >= 64) || s2 != s || b2 != b;
        if (head && !done) {
            vv += v2;
            c += c2;
            mn = fmin(mn, mn2);
            mx = fmax(mx, mx2);
        }
        if (__all(done)) break;
    }
    W.s = s;
    W.b = b;
    W.vv = vv;
    W.mn = mn;
    W.mx = mx;
    W.c = c;
    W.head = head;
}

// The headline fused kernel: decode(PLAIN in place) + ts-range/series-set
// filter + MergeExec dedup + hash group-by aggregate. One workgroup per
// row-group slice, grid-stride; TWO 64-row windows per iteration so two
// dependent table-update chains (L3 probe + atomics) overlap — the kernel
// is memory-latency-bound (SQ_WAIT_ANY 76%) at full occupancy, so the
// lever is per-wave memory-level parallelism.
extern "C" __global__ void __launch_bounds__(256)
k_scan_agg(AggParams P) {
    unsigned long long my_matched = 0;
    const int lane = threadIdx.x & 63;
    for (uint32_t rgi = blockIdx.x; rgi < P.n_rgs; rgi += gridDim.x) {
        // early-abort once the table saturates: the host retries with a
        // larger table, so finishing a doomed pass only burns time.
        // One lane per wave polls the hot counter (same-address loads from
        // every lane would serialize at the coherence point).
        if ((rgi & 7u) == (blockIdx.x & 7u)) {
            // best-effort saturation check: a PLAIN (L1-served) load — a
            // coherent load of this hot word from every wave serializes at
            // the coherence point (~88/us) and cost 50% kernel time; stale
            // reads only delay the abort, which the probe cap bounds anyway
            unsigned long long f = 0;
            if (lane == 0) f = *(volatile const unsigned long long*)P.fill;
            f = __shfl(f, 0, 64);
            if (f > P.fill_limit) {
                if (threadIdx.x == 0)
                    __hip_atomic_fetch_add(P.overflow, 1ull, RLX, AGT);
                break;
            }
        }
        const RgDesc rg = P.rgs[rgi];
        const uint64_t* S = (const uint64_t*)hx_ptr(P.blob, P.dec, rg.series_off);
        const int64_t* T = (const int64_t*)hx_ptr(P.blob, P.dec, rg.ts_off);
        const double* V = (const double*)hx_ptr(P.blob, P.dec, rg.val_off);
        const SstDev sst = P.ssts[rg.sst_id];
        const uint32_t n = rg.n_rows;
        for (uint32_t base = 0; base < n; base += blockDim.x * 2) {
            WinResult A, B;
            B.c = 0;
            B.head = false;
            scan_window(P, rg, sst, S, T, V, base, n, lane, my_matched, A);
            const uint32_t base2 = base + blockDim.x;
            if (base2 < n)
                scan_window(P, rg, sst, S, T, V, base2, n, lane, my_matched, B);
            const bool upA = A.head && A.c > 0;
            const bool upB = B.head && B.c > 0;
            if (upA && upB) {  // two independent chains: overlap them
                agg_update(P, A.s, A.b, A.vv, A.c, A.mn, A.mx, A.hint_i,
                           A.hint_k);
                agg_update(P, B.s, B.b, B.vv, B.c, B.mn, B.mx, B.hint_i,
                           B.hint_k);
            } else if (upA) {
                agg_update(P, A.s, A.b, A.vv, A.c, A.mn, A.mx, A.hint_i,
                           A.hint_k);
            } else if (upB) {
                agg_update(P, B.s, B.b, B.vv, B.c, B.mn, B.mx, B.hint_i,
                           B.hint_k);
            }
        }
    }
    for (int off = 32; off > 0; off >>= 1)
        my_matched += __shfl_down(my_matched, off, 64);
    if ((threadIdx.x & 63) == 0 && my_matched)
        atomicAdd(P.matched, my_matched);
}

// ---------------------------------------------------------------------------
// Gang kernel (series-only grouping): one workgroup aggregates a GANG of
// same-ordinal row groups from many SSTs into an LDS hash table, then
// flushes distinct keys to the global table. Same-ordinal row groups cover
// roughly the same series window (SSTs are PK-sorted over one id universe),
// so a series touched by G SSTs costs ONE global update instead of G —
// global atomic traffic drops by ~G (guideline 12 at gang scale).
// ---------------------------------------------------------------------------
struct GangParams {
    AggParams P;
    uint32_t gang_size;
    uint32_t n_gangs;
    uint32_t ne;        // LDS hash entries (power of two)
    uint32_t has_mm;    // min/max tracked
};

template <bool MM>
__device__ __forceinline__ void lds_update(const AggParams& P, uint64_t* lkey,
                                           double* lsum, unsigned int* lcnt,
                                           unsigned long long* lmin,
                                           unsigned long long* lmax,
                                           uint32_t ne, uint64_t sv, double v,
                                           uint32_t cnt, double mn, double mx,
                                           uint32_t i0 = 0xFFFFFFFFu,
                                           uint64_t pre = 0,
                                           bool has_pre = false) {
    uint32_t i = (i0 != 0xFFFFFFFFu) ? i0 : ((uint32_t)mix64(sv) & (ne - 1));
    // bisect modes (wrong results; cost decomposition only):
    // skip=5 probe-read only, no atomics; skip=6 atomics to the probe slot
    // without key verification
    if (P.skip == 5) {
        if (lkey[i] == sv) return;
        return;
    }
    if (P.skip == 6) {
        atomicAdd(&lsum[i], v);
        atomicAdd(&lcnt[i], cnt);
        return;
    }
#pragma unroll 1
    for (int probes = 0; probes < 16; probes++) {
        // probe prefetch: the caller read slot i0 right after the row loads,
        // so the common-case key check pays no fresh LDS round-trip on the
        // update critical path. A stale EMPTY only re-routes through the
        // CAS (keys transition EMPTY->key exactly once), never corrupts.
        uint64_t kk = (has_pre && probes == 0) ? pre : lkey[i];
        if (kk == KEY_EMPTY) {
            uint64_t old = atomicCAS(&lkey[i], KEY_EMPTY, sv);
            kk = (old == KEY_EMPTY) ? sv : old;
        }
        if (kk == sv) {
            atomicAdd(&lsum[i], v);
            atomicAdd(&lcnt[i], cnt);
            if (MM) {
                atomicMin(&lmin[i], f64_ordered(mn));
                atomicMax(&lmax[i], f64_ordered(mx));
            }
            return;
        }
        i = (i + 1) & (ne - 1);
    }
    // LDS table full: direct global update (count it — a non-trivial rate
    // here turns the kernel into a global-RMW storm; wave-aggregated so the
    // counter itself cannot serialize)
    if (P.fallback) {
        const unsigned long long mk = __ballot(1);
        const int lane = (int)(threadIdx.x & 63);
        if ((__ffsll(mk) - 1) == lane)
            __hip_atomic_fetch_add(P.fallback,
                                   (unsigned long long)__popcll(mk), RLX, AGT);
    }
    agg_update(P, sv, 0, v, cnt, mn, mx);
}

template <bool MM>
__global__ void __launch_bounds__(1024)
k_scan_agg_gang(const GangParams* __restrict__ gp,
                const uint8_t* __restrict__ blob_arg,
                const uint8_t* __restrict__ dec_arg,
                const RgDesc* __restrict__ rgs_arg) {
    // blob/dec/rgs come as KERNEL ARGUMENTS: pointers loaded from memory are
    // generic (flat_load — slow); argument pointers keep the global address
    // space (global_load).
    const GangParams& G = *gp;
    // Param handling: ~12 hot scalars live in SGPRs; everything the cold
    // paths need (table pointers, cluster arrays, counters) is mirrored
    // into LDS once per block — neither SGPR spills (by-value struct) nor
    // per-iteration s_load stalls (pointer chasing).
    const uint8_t* const blob = blob_arg;
    const uint8_t* const dec = dec_arg;
    const int64_t ts_lo = G.P.ts_lo, ts_hi = G.P.ts_hi;
    const int32_t skip = G.P.skip;
    const int32_t use_sset = G.P.use_sset;
    const uint32_t gang_size = G.gang_size;
    const uint32_t n_gangs = G.n_gangs;
    const uint32_t n_rgs_all = G.P.n_rgs;
    const RgDesc* const rgs_all = rgs_arg;
    const SstDev* const ssts_all = G.P.ssts;
    unsigned long long* const matched_ptr = G.P.matched;
    // Transposed gang walk: a thread owns TWO row positions of the aligned
    // series window and visits them across every unit (row-group slice) of
    // the gang. Same-size SSTs slice into EXACTLY aligned units, so the
    // series at a fixed row position is constant across them — the thread
    // accumulates in registers and touches the LDS table only when the key
    // changes. Loads stay coalesced (lanes = consecutive rows) and the
    // unit-loop iterations are independent (memory-level parallelism).
    // Alignment is a speed matter only: register combining merges equal
    // ADJACENT keys, which is correct for any data.
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const uint32_t ne = G.ne;
    AggParams* Pm = (AggParams*)smem;          // LDS mirror for cold paths
    RgDesc* ldesc = (RgDesc*)(smem + 320);
    const size_t desc_bytes = 320 + (size_t)gang_size * sizeof(RgDesc);
    uint64_t* lkey = (uint64_t*)(smem + desc_bytes);
    double* lsum = (double*)(smem + desc_bytes + (size_t)ne * 8);
    unsigned long long* lmin =
        (unsigned long long*)(smem + desc_bytes + (size_t)ne * 16);
    unsigned long long* lmax =
        (unsigned long long*)(smem + desc_bytes + (size_t)ne * (MM ? 24 : 16));
    unsigned int* lcnt =
        (unsigned int*)(smem + desc_bytes + (size_t)ne * (MM ? 32 : 16));
    {
        static_assert(sizeof(AggParams) <= 320, "grow the LDS mirror");
        const uint64_t* src = (const uint64_t*)&gp->P;
        uint64_t* dst = (uint64_t*)Pm;
        for (uint32_t i = threadIdx.x; i < sizeof(AggParams) / 8;
             i += blockDim.x)
            dst[i] = src[i];
    }

    const int lane = threadIdx.x & 63;
    unsigned long long my_matched = 0;
    unsigned long long my_flushes = 0;
    for (uint32_t gang = blockIdx.x; gang < n_gangs; gang += gridDim.x) {
        const uint32_t rg0 = gang * gang_size;
        const uint32_t rg_end = min(rg0 + gang_size, n_rgs_all);
        const uint32_t nu = rg_end - rg0;
        for (uint32_t i = threadIdx.x; i < ne; i += blockDim.x) {
            lkey[i] = KEY_EMPTY;
            lsum[i] = 0.0;
            lcnt[i] = 0u;
            if (MM) {
                lmin[i] = ~0ull;
                lmax[i] = 0ull;
            }
        }
        // preload unit descriptors (one dynamic-LDS object only — G17)
        for (uint32_t u = threadIdx.x; u < nu; u += blockDim.x)
            ldesc[u] = rgs_all[rg0 + u];
        __syncthreads();
        uint32_t max_n = 0;
        for (uint32_t u = 0; u < nu; u++) max_n = max(max_n, ldesc[u].n_rows);

        bool have_run = false;
        uint64_t run_key = 0;
        double run_sum = 0, run_min = 0, run_max = 0;
        uint32_t run_cnt = 0;
        for (uint32_t roff = 0; roff < max_n; roff += blockDim.x) {
            const uint32_t r = roff + threadIdx.x;
            // two-stage software pipeline over the units: issue unit u+1's
            // column loads before processing unit u, so each iteration's
            // dependent-load stall overlaps the next iteration's fetches
            // (the compiler alone serializes them behind per-iteration
            // waits — measured ~730 cy/iteration without this).
            uint64_t sA = KEY_EMPTY, s1A = KEY_EMPTY;
            int64_t tA = 0, t1A = 0;
            double vA = 0;
            uint32_t nA = 0, sstA = 0;
            int32_t nxA = -1;
            auto issue = [&](uint32_t u, uint64_t& s_, uint64_t& s1_,
                             int64_t& t_, int64_t& t1_, double& v_,
                             uint32_t& n_, uint32_t& sst_, int32_t& nx_) {
                const RgDesc rg = ldesc[u];
                n_ = rg.n_rows;
                sst_ = rg.sst_id;
                nx_ = rg.next_rg;
                const bool inb = r < n_;
                const bool inb1 = r + 1 < n_;
                const uint64_t* S =
                    (const uint64_t*)hx_ptr(blob, dec, rg.series_off);
                const int64_t* T =
                    (const int64_t*)hx_ptr(blob, dec, rg.ts_off);
                const double* V =
                    (const double*)hx_ptr(blob, dec, rg.val_off);
                s_ = inb ? S[r] : KEY_EMPTY;
                t_ = inb ? T[r] : 0;
                v_ = inb ? V[r] : 0.0;
                s1_ = inb1 ? S[r + 1] : KEY_EMPTY;
                t1_ = inb1 ? T[r + 1] : 0;
            };
            if (nu > 0)
                issue(0, sA, s1A, tA, t1A, vA, nA, sstA, nxA);
#pragma unroll 1
            for (uint32_t u = 0; u < nu; u++) {
                uint64_t sB = KEY_EMPTY, s1B = KEY_EMPTY;
                int64_t tB = 0, t1B = 0;
                double vB = 0;
                uint32_t nB = 0, sstB = 0;
                int32_t nxB = -1;
                if (u + 1 < nu)
                    issue(u + 1, sB, s1B, tB, t1B, vB, nB, sstB, nxB);
                const uint64_t sv = sA;
                const int64_t tv = tA;
                if (r < nA) {
                    bool alive = (tv >= ts_lo) & (tv < ts_hi);
                    if (alive && use_sset) alive = sset_has(*Pm, sv);
                    if (alive && skip < 2) {
                        bool dup = false;
                        if (r + 1 < nA) {
                            dup = (s1A == sv) & (t1A == tv);
                        } else if (nxA >= 0) {
                            const RgDesc nx = rgs_all[nxA];
                            uint64_t s2 = *(const uint64_t*)hx_ptr(
                                blob, dec, nx.series_off);
                            int64_t t2 = *(const int64_t*)hx_ptr(blob, dec,
                                                                 nx.ts_off);
                            dup = (s2 == sv) & (t2 == tv);
                        }
                        if (!dup) {
                            const SstDev sst = ssts_all[sstA];
                            if (sst.cluster >= 0) {
                                uint64_t msq = sst.dense_seq
                                    ? ((const uint64_t*)(dec +
                                           (sst.dense_seq & OFF_MASK)))
                                          [ldesc[u].row_base + r]
                                    : sst.seq;
                                dup = shadowed(*Pm, sst, msq, sv, tv);
                            }
                        }
                        alive = !dup;
                    }
                    if (alive) {
                        const double v = vA;
                        my_matched++;
                        if (skip == 1 || skip == 2) {
                        } else if (have_run && run_key == sv) {
                            run_sum += v;
                            run_cnt++;
                            if (MM) {
                                run_min = fmin(run_min, v);
                                run_max = fmax(run_max, v);
                            }
                        } else {
                            if (have_run) {
                                if (skip == 3) my_flushes++;  // bisect
                                else lds_update<MM>(*Pm, lkey, lsum, lcnt,
                                                    lmin, lmax, ne, run_key,
                                                    run_sum, run_cnt, run_min,
                                                    run_max);
                            }
                            have_run = true;
                            run_key = sv;
                            run_sum = v;
                            run_cnt = 1;
                            run_min = v;
                            run_max = v;
                        }
                    }
                }
                sA = sB; s1A = s1B; tA = tB; t1A = t1B; vA = vB;
                nA = nB; sstA = sstB; nxA = nxB;
            }
        }
        if (have_run) {
            lds_update<MM>(*Pm, lkey, lsum, lcnt, lmin, lmax, ne, run_key,
                           run_sum, run_cnt, run_min, run_max);
            have_run = false;
        }
        __syncthreads();
        for (uint32_t i = threadIdx.x; i < ne; i += blockDim.x) {
            if (lkey[i] == KEY_EMPTY) continue;
            agg_update(*Pm, lkey[i], 0, lsum[i], (unsigned long long)lcnt[i],
                       MM ? ordered_f64(lmin[i]) : 0.0,
                       MM ? ordered_f64(lmax[i]) : 0.0);
        }
        __syncthreads();
    }
    if (skip == 3) my_matched = my_flushes;  // bisect: report flush count
    for (int off = 32; off > 0; off >>= 1)
        my_matched += __shfl_down(my_matched, off, 64);
    if ((threadIdx.x & 63) == 0 && my_matched)
        atomicAdd(matched_ptr, my_matched);
}

// ---------------------------------------------------------------------------
// Series-range partitioned aggregation. Position-aligned LDS
// gangs lose to series-position jitter (sqrt-scale fluctuations between
// SSTs dwarf a window), so the partition here is by series VALUE: block b
// owns [bounds[b], bounds[b+1]) across every SST; its series fit an LDS
// table by construction, flushed once — global RMWs drop from one per
// window-run (~rows/1.6) to one per (block, series).
// ---------------------------------------------------------------------------

// Stride-512 series samples per rg slice (16 fixed slots, ~0 padding): the
// host sorts these into equal-sample quantile boundaries. The stride phase
// is JITTERED per slice (deterministic hash of the slice index): SSTs of one
// store share the same sorted series universe, so un-jittered samples land
// on near-identical series in every SST and the distinct-count solve
// under-estimates by ~the SST count (measured 39k for 10M at the 1B shape —
// the round-1 LDS-fallback storm). Jitter makes samples ~independent row
// draws, which is exactly the model the u = x(1-e^{-m/x}) solver assumes.
extern "C" __global__ void __launch_bounds__(256)
k_sample_series(const RgDesc* __restrict__ rgs, uint32_t n_rgs,
                const uint8_t* __restrict__ blob,
                const uint8_t* __restrict__ dec, uint64_t* __restrict__ out) {
    const int lane = threadIdx.x & 63;
    const uint32_t wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const uint32_t n_waves = (gridDim.x * blockDim.x) >> 6;
    for (uint32_t g = wave_id; g < n_rgs; g += n_waves) {
        const RgDesc rg = rgs[g];
        const uint64_t* S =
            (const uint64_t*)hx_ptr(blob, dec, rg.series_off);
        const uint32_t phase =
            (uint32_t)(mix64((uint64_t)g ^ 0x9E3779B97F4A7C15ull) >> 40) &
            511u;
        if (lane < 16) {
            uint32_t r = phase + (uint32_t)lane * 512u;
            out[(size_t)g * 16 + lane] = (r < rg.n_rows) ? S[r] : ~0ull;
        }
    }
}

// Wave-parallel 64-ary lower_bound: count of indices in [0,n) whose probed
// key is < target (keys ascending). ~log64(n) rounds of one parallel load.
template <typename Pred>
__device__ __forceinline__ uint32_t lb64(uint32_t n, uint64_t target,
                                         int lane, Pred key_at) {
    uint32_t lo = 0, rem = n;
    while (rem > 0) {
        const uint32_t step = (rem + 63u) >> 6;
        const uint32_t p = lo + (uint32_t)lane * step;
        const bool in = (uint32_t)lane * step < rem;
        const bool t = in && (key_at(p) < target);
        const uint32_t k = (uint32_t)__popcll(__ballot(t));
        if (k == 0) break;
        const uint32_t adv = (k - 1) * step + 1;
        lo += adv;
        rem = (step - 1u) < (rem - adv) ? (step - 1u) : (rem - adv);
    }
    return lo;   // first index with key >= target (== n if none)
}

// Row bounds per (boundary, sst): first staged row with series >= bounds[b],
// packed (rg-list position << 32 | row within slice). Computed once per
// prepare (decode is deterministic, so bounds survive re-decode).
extern "C" __global__ void __launch_bounds__(256)
k_range_bounds(AggParams P, RangeAux R, uint64_t* __restrict__ out) {
    const int lane = threadIdx.x & 63;
    const uint32_t wave_id = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const uint32_t n_waves = (gridDim.x * blockDim.x) >> 6;
    const uint32_t total = (R.n_blocks + 1) * R.n_ssts;
    for (uint32_t w = wave_id; w < total; w += n_waves) {
        const uint32_t b = w / R.n_ssts, si = w % R.n_ssts;
        const uint64_t target = R.bounds[b];
        const int32_t off = R.sst_rg_off[si];
        const uint32_t cnt = (uint32_t)R.sst_rg_cnt[si];
        // level 1: over slices by their first-row series
        const uint32_t g = lb64(cnt, target, lane, [&](uint32_t p) {
            const RgDesc rg = P.rgs[R.sst_rgs[off + (int32_t)p]];
            return *(const uint64_t*)hx_ptr(P.blob, P.dec, rg.series_off);
        });
        uint64_t packed;
        if (g == 0) {
            packed = 0;   // boundary at the very first row
        } else {
            // boundary row is inside slice g-1, or at the start of slice g
            const RgDesc rg = P.rgs[R.sst_rgs[off + (int32_t)(g - 1)]];
            const uint64_t* S =
                (const uint64_t*)hx_ptr(P.blob, P.dec, rg.series_off);
            const uint32_t r = lb64(rg.n_rows, target, lane,
                                    [&](uint32_t p) { return S[p]; });
            packed = (r == rg.n_rows) ? ((uint64_t)g << 32)
                                      : (((uint64_t)(g - 1) << 32) | r);
        }
        if (lane == 0) out[w] = packed;
    }
}

// 64-row window worker for the range kernel: like scan_window but per-wave
// (lane-indexed rows), clamped to [base, hi) with successor dedup against
// the slice's TRUE row count, and no global-table probe hint. Templated on
// MM so the sum/count path never pays the min/max cross-lane shuffles
// (each 64-bit __shfl is two ds_bpermutes — real LDS-array cycles).
template <bool MM>
__device__ __forceinline__ void scan_window_range(
    const AggParams& P, const RgDesc& rg, const SstDev& sst,
    const uint64_t* S, const int64_t* T, const double* V, uint32_t base,
    uint32_t hi, uint32_t n_true, int lane, unsigned long long& my_matched,
    WinResult& W, const uint64_t* lkey, uint32_t ne, uint32_t interp,
    uint64_t ilo, double islope) {
    const uint32_t r = base + (uint32_t)lane;
    const bool inb = r < hi;
    const bool has_next = r + 1 < n_true;
    uint64_t s = KEY_EMPTY;
    int64_t t = 0;
    double v = 0.0;
    uint64_t s1 = 0;
    int64_t t1 = 0;
    bool alive = false;
    if (inb) {
        t = T[r];
        s = S[r];
        v = V[r];
        if (has_next) {
            s1 = S[r + 1];
            t1 = T[r + 1];
        }
    }
    // probe prefetch: the head lane's LDS slot is known from `s` alone —
    // issue the probe read NOW so its latency hides under filter/dedup
    {
        uint32_t slot;
        if (interp) {
            uint32_t x = (uint32_t)((double)(s - ilo) * islope);
            slot = x < ne ? x : ne - 1;
        } else {
            slot = (uint32_t)mix64(s) & (ne - 1);
        }
        W.hint_i = slot;
        W.hint_k = lkey[slot];
    }
    if (inb) {
        alive = (t >= P.ts_lo) & (t < P.ts_hi);
        if (alive && P.use_sset) alive = sset_has(P, s);
        if (alive && P.skip < 2) {
            bool dup = false;
            if (has_next) {
                dup = (s1 == s) & (t1 == t);
            } else if (rg.next_rg >= 0) {
                const RgDesc nx = P.rgs[rg.next_rg];
                uint64_t s2 = *(const uint64_t*)hx_ptr(P.blob, P.dec,
                                                       nx.series_off);
                int64_t t2 = *(const int64_t*)hx_ptr(P.blob, P.dec, nx.ts_off);
                dup = (s2 == s) & (t2 == t);
            }
            if (!dup && sst.cluster >= 0)
            dup = shadowed(P, sst, row_seq(P, sst, rg, r), s, t);
            alive = !dup;
        }
        if (!alive) v = 0.0;
    }
    unsigned long long c = alive ? 1ull : 0ull;
    double vv = alive ? v : 0.0;
    double mn = alive ? v : HUGE_VAL;
    double mx = alive ? v : -HUGE_VAL;
    my_matched += c;
    if (P.skip == 4) {
        // bisect mode: no wave pre-reduction — every alive lane updates the
        // LDS table itself (short runs make the shfl loop VALU-heavy)
        W.s = s;
        W.b = 0;
        W.vv = vv;
        W.mn = mn;
        W.mx = mx;
        W.c = c;
        W.head = alive;
        return;
    }
    const uint64_t sp = __shfl_up(s, 1, 64);
    const bool head = (lane == 0) || sp != s;
    bool done = false;
    for (int d = 1; d < 64; d++) {
        const uint64_t s2 = __shfl_down(s, d, 64);
        const double v2 = __shfl_down(vv, d, 64);
        const unsigned long long c2 = __shfl_down(c, d, 64);
        const double mn2 = MM ? __shfl_down(mn, d, 64) : 0.0;
        const double mx2 = MM ? __shfl_down(mx, d, 64) : 0.0;
        done = done || (lane + d >= 64) || s2 != s;
        if (head && !done) {
            vv += v2;
            c += c2;
            if (MM) {
                mn = fmin(mn, mn2);
                mx = fmax(mx, mx2);
            }
        }
        if (__all(done)) break;
    }
    W.s = s;
    W.b = 0;
    W.vv = vv;
    W.mn = mn;
    W.mx = mx;
    W.c = c;
    W.head = head;
}

template <bool MM, int MINW>
__global__ void __launch_bounds__(256, MINW)
k_scan_agg_range(AggParams P, RangeAux R) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    __shared__ int s_abort;   // global table saturated: stop, host retries
    const uint32_t ne = R.ne;
    uint64_t* lkey = (uint64_t*)smem;
    double* lsum = (double*)(smem + (size_t)ne * 8);
    unsigned long long* lmin = (unsigned long long*)(smem + (size_t)ne * 16);
    unsigned long long* lmax = (unsigned long long*)(smem + (size_t)ne * 24);
    unsigned int* lcnt = (unsigned int*)(smem + (size_t)ne * (MM ? 32 : 16));
    const int lane = threadIdx.x & 63;
    const uint32_t wave = threadIdx.x >> 6;
    const uint32_t n_waves = blockDim.x >> 6;
    unsigned long long my_matched = 0;
    for (uint32_t blk = blockIdx.x; blk < R.n_blocks; blk += gridDim.x) {
        // saturation early-abort, ONCE PER BLOCK: P.fill's line is evicted
        // chip-wide by every agent-scope claim, so ANY load of it is served
        // at the coherence point (~88/us single-word) — polling per
        // (wave, sst) serialized the whole kernel (+23 ms at the 1B shape).
        // A stale read only delays the abort; the probe cap still bounds a
        // doomed pass.
        if (threadIdx.x == 0) {
            s_abort = 0;
            if (P.poll && (blk & 7u) == 0) {
                unsigned long long f = __hip_atomic_load(
                    P.fill, RLX, __HIP_MEMORY_SCOPE_WORKGROUP);
                if (f > P.fill_limit) s_abort = 1;
            }
        }
        for (uint32_t i = threadIdx.x; i < ne; i += blockDim.x) {
            lkey[i] = KEY_EMPTY;
            lsum[i] = 0.0;
            lcnt[i] = 0u;
            if (MM) {
                lmin[i] = ~0ull;
                lmax[i] = 0ull;
            }
        }
        // interpolation slotting (R.interp): first probe at the key's
        // fractional position within the block's series range. Heads arrive
        // in ascending series order, so a wave's DS ops hit consecutive
        // slots = consecutive banks (conflict-free); collisions fall back to
        // linear probing as before.
        double islope = 0.0;
        uint64_t ilo = 0;
        if (R.interp) {
            ilo = R.bounds[blk];
            const uint64_t ihi = R.bounds[blk + 1];
            islope = (double)ne / ((double)(ihi - ilo) + 1.0);
        }
        __syncthreads();
        if (s_abort) {
            if (threadIdx.x == 0)
                __hip_atomic_fetch_add(P.overflow, 1ull, RLX, AGT);
            break;   // host retries with a larger table
        }
        for (uint32_t si = wave; si < R.n_ssts; si += n_waves) {
            const int32_t loff = R.sst_rg_off[si];
            const uint64_t pk0 = R.bound_rows[(size_t)blk * R.n_ssts + si];
            const uint64_t pk1 =
                R.bound_rows[(size_t)(blk + 1) * R.n_ssts + si];
            uint32_t pos = (uint32_t)(pk0 >> 32);
            uint32_t row = (uint32_t)pk0;
            const uint32_t epos = (uint32_t)(pk1 >> 32);
            const uint32_t erow = (uint32_t)pk1;
            while (pos < epos || (pos == epos && row < erow)) {
                const RgDesc rg = P.rgs[R.sst_rgs[loff + (int32_t)pos]];
                const uint64_t* S =
                    (const uint64_t*)hx_ptr(P.blob, P.dec, rg.series_off);
                const int64_t* T =
                    (const int64_t*)hx_ptr(P.blob, P.dec, rg.ts_off);
                const double* V =
                    (const double*)hx_ptr(P.blob, P.dec, rg.val_off);
                const SstDev sst = P.ssts[rg.sst_id];
                const uint32_t hi = (pos == epos) ? erow : rg.n_rows;
                // dual 64-row windows: two independent load->reduce->LDS
                // chains in flight per wave (quad measured equal-to-worse:
                // the bound is not window-chain MLP)
                for (uint32_t base = row; base < hi; base += 128) {
                    WinResult A, B;
                    B.c = 0;
                    B.head = false;
                    scan_window_range<MM>(P, rg, sst, S, T, V, base, hi,
                                          rg.n_rows, lane, my_matched, A,
                                          lkey, ne, R.interp, ilo, islope);
                    if (base + 64 < hi)
                        scan_window_range<MM>(P, rg, sst, S, T, V, base + 64,
                                              hi, rg.n_rows, lane, my_matched,
                                              B, lkey, ne, R.interp, ilo,
                                              islope);
                    if (P.skip == 1) continue;
                    // verified-hit short circuit: the probe read happened at
                    // load time; hint_k == key PROVES the slot (keys are
                    // write-once), so the adds go straight to LDS with no
                    // fresh read and no probe-loop control flow — the loop
                    // structure alone measured ~19.5 ms of the 23.9 ms
                    // kernel (skip6 bisect)
                    if (A.head && A.c > 0) {
                        if (A.hint_k == A.s) {
                            atomicAdd(&lsum[A.hint_i], A.vv);
                            atomicAdd(&lcnt[A.hint_i], (uint32_t)A.c);
                            if (MM) {
                                atomicMin(&lmin[A.hint_i], f64_ordered(A.mn));
                                atomicMax(&lmax[A.hint_i], f64_ordered(A.mx));
                            }
                        } else {
                            lds_update<MM>(P, lkey, lsum, lcnt, lmin, lmax,
                                           ne, A.s, A.vv, (uint32_t)A.c,
                                           A.mn, A.mx, A.hint_i);
                        }
                    }
                    if (B.head && B.c > 0) {
                        if (B.hint_k == B.s) {
                            atomicAdd(&lsum[B.hint_i], B.vv);
                            atomicAdd(&lcnt[B.hint_i], (uint32_t)B.c);
                            if (MM) {
                                atomicMin(&lmin[B.hint_i], f64_ordered(B.mn));
                                atomicMax(&lmax[B.hint_i], f64_ordered(B.mx));
                            }
                        } else {
                            lds_update<MM>(P, lkey, lsum, lcnt, lmin, lmax,
                                           ne, B.s, B.vv, (uint32_t)B.c,
                                           B.mn, B.mx, B.hint_i);
                        }
                    }
                }
                pos++;
                row = 0;
            }
        }
        __syncthreads();
        {
            unsigned long long live = 0;
            for (uint32_t i = threadIdx.x; i < ne; i += blockDim.x) {
                if (lkey[i] == KEY_EMPTY) continue;
                live++;
                agg_update(P, lkey[i], 0, lsum[i],
                           (unsigned long long)lcnt[i],
                           MM ? ordered_f64(lmin[i]) : 0.0,
                           MM ? ordered_f64(lmax[i]) : 0.0, 0xFFFFFFFFu, 0,
                           false);
            }
            for (int off = 32; off > 0; off >>= 1)
                live += __shfl_down(live, off, 64);
            if ((threadIdx.x & 63) == 0 && live)
                __hip_atomic_fetch_add(P.fill, live, RLX, AGT);
        }
        __syncthreads();
    }
    for (int off = 32; off > 0; off >>= 1)
        my_matched += __shfl_down(my_matched, off, 64);
    if ((threadIdx.x & 63) == 0 && my_matched)
        atomicAdd(P.matched, my_matched);
}

// ---------------------------------------------------------------------------
// Pair-load series-range kernel (round-2 redesign of k_scan_agg_range).
// The r01 kernel was 95% SQ_WAIT_ANY at ~2% LDS-array utilization: pure
// latency exposure, not throughput. This variant removes latency sources:
//   - 16 B/lane dwordx4 loads (the calibrated wide-stream width): each lane
//     owns TWO consecutive rows, halving load instructions per row;
//   - the successor row for dedup comes from the register pair / one
//     cross-lane shuffle instead of a second global read of S and T;
//   - NO cross-lane run pre-reduction (measured ±3%): each lane merges its
//     own pair in registers (36% of adjacent rows share a series at the
//     headline shape) and issues its 1-2 LDS updates directly — the shuffle
//     loop's ~7 dependent ds_bpermute rounds per window disappear.
// Table layout, partitioning, flush and dedup semantics are identical to
// k_scan_agg_range.
// ---------------------------------------------------------------------------
template <bool MM>
__global__ void __launch_bounds__(256)
k_scan_agg_range2(AggParams P, RangeAux R) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    __shared__ int s_abort;
    const uint32_t ne = R.ne;
    uint64_t* lkey = (uint64_t*)smem;
    double* lsum = (double*)(smem + (size_t)ne * 8);
    unsigned long long* lmin = (unsigned long long*)(smem + (size_t)ne * 16);
    unsigned long long* lmax = (unsigned long long*)(smem + (size_t)ne * 24);
    unsigned int* lcnt = (unsigned int*)(smem + (size_t)ne * (MM ? 32 : 16));
    const int lane = threadIdx.x & 63;
    const uint32_t wave = threadIdx.x >> 6;
    const uint32_t n_waves = blockDim.x >> 6;
    unsigned long long my_matched = 0;
    for (uint32_t blk = blockIdx.x; blk < R.n_blocks; blk += gridDim.x) {
        if (threadIdx.x == 0) {
            s_abort = 0;
            if (P.poll && (blk & 7u) == 0) {
                unsigned long long f = __hip_atomic_load(
                    P.fill, RLX, __HIP_MEMORY_SCOPE_WORKGROUP);
                if (f > P.fill_limit) s_abort = 1;
            }
        }
        for (uint32_t i = threadIdx.x; i < ne; i += blockDim.x) {
            lkey[i] = KEY_EMPTY;
            lsum[i] = 0.0;
            lcnt[i] = 0u;
            if (MM) {
                lmin[i] = ~0ull;
                lmax[i] = 0ull;
            }
        }
        double islope = 0.0;
        uint64_t ilo = 0;
        if (R.interp) {
            ilo = R.bounds[blk];
            const uint64_t ihi = R.bounds[blk + 1];
            islope = (double)ne / ((double)(ihi - ilo) + 1.0);
        }
        __syncthreads();
        if (s_abort) {
            if (threadIdx.x == 0)
                __hip_atomic_fetch_add(P.overflow, 1ull, RLX, AGT);
            break;
        }
        for (uint32_t si = wave; si < R.n_ssts; si += n_waves) {
            const int32_t loff = R.sst_rg_off[si];
            const uint64_t pk0 = R.bound_rows[(size_t)blk * R.n_ssts + si];
            const uint64_t pk1 =
                R.bound_rows[(size_t)(blk + 1) * R.n_ssts + si];
            uint32_t pos = (uint32_t)(pk0 >> 32);
            uint32_t row = (uint32_t)pk0;
            const uint32_t epos = (uint32_t)(pk1 >> 32);
            const uint32_t erow = (uint32_t)pk1;
            while (pos < epos || (pos == epos && row < erow)) {
                const RgDesc rg = P.rgs[R.sst_rgs[loff + (int32_t)pos]];
                const uint64_t* S =
                    (const uint64_t*)hx_ptr(P.blob, P.dec, rg.series_off);
                const int64_t* T =
                    (const int64_t*)hx_ptr(P.blob, P.dec, rg.ts_off);
                const double* V =
                    (const double*)hx_ptr(P.blob, P.dec, rg.val_off);
                const SstDev sst = P.ssts[rg.sst_id];
                const uint32_t hi = (pos == epos) ? erow : rg.n_rows;
                const uint32_t nt = rg.n_rows;   // true slice rows
                for (uint32_t base = row & ~1u; base < hi; base += 128) {
                    const uint32_t r0 = base + 2u * (uint32_t)lane;
                    const uint32_t r1 = r0 + 1u;
                    // load the pair (dwordx4); mask by the ARRAY bound nt —
                    // rows in [hi, nt) load safely and are filtered below
                    uint64_t s0 = KEY_EMPTY, s1 = KEY_EMPTY;
                    int64_t t0 = 0, t1 = 0;
                    double v0 = 0.0, v1 = 0.0;
                    if (r1 < nt) {
                        const ulonglong2 sp =
                            *(const ulonglong2*)(S + r0);
                        const longlong2 tp = *(const longlong2*)(T + r0);
                        const double2 vp = *(const double2*)(V + r0);
                        s0 = sp.x;
                        s1 = sp.y;
                        t0 = tp.x;
                        t1 = tp.y;
                        v0 = vp.x;
                        v1 = vp.y;
                    } else if (r0 < nt) {
                        s0 = S[r0];
                        t0 = T[r0];
                        v0 = V[r0];
                    }
                    // slot + probe prefetch: issue the LDS probe reads NOW
                    // so their ~50-cycle latency hides under filter/dedup
                    uint32_t iA, iB;
                    if (R.interp) {
                        uint32_t x0 = (uint32_t)((double)(s0 - ilo) * islope);
                        uint32_t x1 = (uint32_t)((double)(s1 - ilo) * islope);
                        iA = x0 < ne ? x0 : ne - 1;
                        iB = x1 < ne ? x1 : ne - 1;
                    } else {
                        iA = (uint32_t)mix64(s0) & (ne - 1);
                        iB = (uint32_t)mix64(s1) & (ne - 1);
                    }
                    const uint64_t preA = lkey[iA];
                    const uint64_t preB = lkey[iB];
                    // window validity: [max(row, base), hi)
                    bool a0 = r0 >= row && r0 < hi;
                    bool a1 = r1 >= row && r1 < hi;
                    a0 = a0 && (t0 >= P.ts_lo) & (t0 < P.ts_hi);
                    a1 = a1 && (t1 >= P.ts_lo) & (t1 < P.ts_hi);
                    if (P.use_sset) {
                        if (a0) a0 = sset_has(P, s0);
                        if (a1) a1 = sset_has(P, s1);
                    }
                    // successor of r1 = next lane's r0 (one 64-bit shuffle
                    // pair); lane 63 / slice-tail fall back to scalar loads
                    const uint64_t sn = __shfl_down(s0, 1, 64);
                    const int64_t tn = __shfl_down((long long)t0, 1, 64);
                    if (a0 && P.skip < 2) {
                        bool dup;
                        if (r1 < nt) {
                            dup = (s1 == s0) & (t1 == t0);
                        } else if (rg.next_rg >= 0) {
                            const RgDesc nx = P.rgs[rg.next_rg];
                            dup = (*(const uint64_t*)hx_ptr(
                                       P.blob, P.dec, nx.series_off) == s0) &
                                  (*(const int64_t*)hx_ptr(
                                       P.blob, P.dec, nx.ts_off) == t0);
                        } else {
                            dup = false;
                        }
                        if (!dup && sst.cluster >= 0)
                            dup = shadowed(P, sst, row_seq(P, sst, rg, r0),
                                           s0, t0);
                        a0 = !dup;
                    }
                    if (a1 && P.skip < 2) {
                        bool dup;
                        const uint32_t r2 = r1 + 1u;
                        if (r2 < nt) {
                            if (lane < 63) {
                                dup = (sn == s1) & (tn == t1);
                            } else {
                                dup = (S[r2] == s1) & (T[r2] == t1);
                            }
                        } else if (rg.next_rg >= 0) {
                            const RgDesc nx = P.rgs[rg.next_rg];
                            dup = (*(const uint64_t*)hx_ptr(
                                       P.blob, P.dec, nx.series_off) == s1) &
                                  (*(const int64_t*)hx_ptr(
                                       P.blob, P.dec, nx.ts_off) == t1);
                        } else {
                            dup = false;
                        }
                        if (!dup && sst.cluster >= 0)
                            dup = shadowed(P, sst, row_seq(P, sst, rg, r1),
                                           s1, t1);
                        a1 = !dup;
                    }
                    my_matched += (a0 ? 1u : 0u) + (a1 ? 1u : 0u);
                    if (P.skip == 1) continue;
                    // in-lane pair merge, then direct LDS updates
                    const bool merge = a0 && a1 && (s0 == s1);
                    if (merge) {
                        lds_update<MM>(P, lkey, lsum, lcnt, lmin, lmax, ne,
                                       s0, v0 + v1, 2u, fmin(v0, v1),
                                       fmax(v0, v1), iA, preA, true);
                    } else {
                        if (a0)
                            lds_update<MM>(P, lkey, lsum, lcnt, lmin, lmax,
                                           ne, s0, v0, 1u, v0, v0, iA, preA,
                                           true);
                        if (a1)
                            lds_update<MM>(P, lkey, lsum, lcnt, lmin, lmax,
                                           ne, s1, v1, 1u, v1, v1, iB, preB,
                                           true);
                    }
                }
                pos++;
                row = 0;
            }
        }
        __syncthreads();
        {
            unsigned long long live = 0;
            for (uint32_t i = threadIdx.x; i < ne; i += blockDim.x) {
                if (lkey[i] == KEY_EMPTY) continue;
                live++;
                agg_update(P, lkey[i], 0, lsum[i],
                           (unsigned long long)lcnt[i],
                           MM ? ordered_f64(lmin[i]) : 0.0,
                           MM ? ordered_f64(lmax[i]) : 0.0, 0xFFFFFFFFu, 0,
                           false);
            }
            for (int off = 32; off > 0; off >>= 1)
                live += __shfl_down(live, off, 64);
            if ((threadIdx.x & 63) == 0 && live)
                __hip_atomic_fetch_add(P.fill, live, RLX, AGT);
        }
        __syncthreads();
    }
    for (int off = 32; off > 0; off >>= 1)
        my_matched += __shfl_down(my_matched, off, 64);
    if ((threadIdx.x & 63) == 0 && my_matched)
        atomicAdd(P.matched, my_matched);
}

// ---------------------------------------------------------------------------
// RLE_DICTIONARY decode (parquet-format Encodings.md "RLE/bit-packed
// hybrid"; the encoding config.rs:54-75 enables with dictionaries on).
// One workgroup per page: lane-serial run-header walk into LDS (varints are
// sequential), then threads expand runs in parallel through the dictionary.
// ---------------------------------------------------------------------------
#define RLED_MAX_RUNS 8192

extern "C" __global__ void __launch_bounds__(256)
k_decode_rledict(const uint8_t* __restrict__ blob, uint8_t* __restrict__ dec,
                 const RleDictPageDesc* __restrict__ pages, uint32_t n_pages,
                 unsigned long long* err_flag) {
    struct Run { uint32_t out_pos, count, byte_off, rle_val_or_flag; };
    __shared__ Run runs[RLED_MAX_RUNS];
    __shared__ int n_runs;
    __shared__ int hdr_err;
    __shared__ uint32_t bw_sh;
    for (uint32_t pg = blockIdx.x; pg < n_pages; pg += gridDim.x) {
        const RleDictPageDesc pd = pages[pg];
        const uint8_t* idx = ((pd.idx_off & OFF_DEC) ? dec : blob) +
                             (pd.idx_off & OFF_MASK);
        const uint64_t* dict = (const uint64_t*)(((pd.dict_off & OFF_DEC)
                                                      ? dec : blob) +
                                                 (pd.dict_off & OFF_MASK));
        uint64_t* dst = (uint64_t*)(dec + (pd.dst_off & OFF_MASK));
        if (threadIdx.x == 0) {
            hdr_err = 0;
            n_runs = 0;
            const uint32_t len = pd.idx_len;
            uint32_t pos = 0;
            uint32_t bw = (len > 0) ? idx[pos++] : 0xFF;
            bw_sh = bw;
            if (bw > 32) hdr_err = 1;
            uint32_t out = 0;
            while (!hdr_err && out < pd.n_values) {
                // varint header
                uint64_t hdrv = 0;
                int sh = 0;
                for (;;) {
                    if (pos >= len || sh > 28) { hdr_err = 2; break; }
                    uint8_t bb = idx[pos++];
                    hdrv |= (uint64_t)(bb & 0x7f) << sh;
                    if (!(bb & 0x80)) break;
                    sh += 7;
                }
                if (hdr_err) break;
                if (n_runs >= RLED_MAX_RUNS) { hdr_err = 3; break; }
                if (hdrv & 1) {  // bit-packed: (hdr>>1) groups of 8
                    uint32_t cnt = (uint32_t)(hdrv >> 1) * 8;
                    uint32_t nbytes = (uint32_t)(hdrv >> 1) * bw;
                    if (pos + nbytes > len) { hdr_err = 4; break; }
                    if (cnt > pd.n_values - out) cnt = pd.n_values - out;
                    runs[n_runs++] = {out, cnt, pos, 0x80000000u};
                    pos += nbytes;
                    out += cnt;
                } else {          // RLE run
                    uint32_t cnt = (uint32_t)(hdrv >> 1);
                    uint32_t vbytes = (bw + 7) / 8;
                    if (pos + vbytes > len) { hdr_err = 5; break; }
                    uint32_t v = 0;
                    for (uint32_t i = 0; i < vbytes; i++)
                        v |= (uint32_t)idx[pos + i] << (8 * i);
                    pos += vbytes;
                    if (cnt > pd.n_values - out) cnt = pd.n_values - out;
                    runs[n_runs++] = {out, cnt, 0, v};
                    out += cnt;
                }
            }
            if (!hdr_err && out < pd.n_values) hdr_err = 6;
        }
        __syncthreads();
        if (hdr_err) {
            if (threadIdx.x == 0) atomicAdd(err_flag, 1ull);
            __syncthreads();
            continue;
        }
        const uint32_t bw = bw_sh;
        const int nr = n_runs;
        for (int rI = 0; rI < nr; rI++) {
            const Run run = runs[rI];
            if (run.rle_val_or_flag != 0x80000000u) {  // RLE: broadcast
                const uint32_t v = run.rle_val_or_flag;
                if (v >= pd.dict_n) {
                    if (threadIdx.x == 0) atomicAdd(err_flag, 1ull);
                    break;
                }
                const uint64_t dv = dict[v];
                for (uint32_t i = threadIdx.x; i < run.count; i += blockDim.x)
                    dst[run.out_pos + i] = dv;
            } else {  // bit-packed: value i at bit i*bw
                bool bad = false;
                for (uint32_t i = threadIdx.x; i < run.count; i += blockDim.x) {
                    const uint64_t bitpos = (uint64_t)i * bw;
                    const uint8_t* base = idx + run.byte_off + (bitpos >> 3);
                    const int shift = (int)(bitpos & 7);
                    uint64_t window = 0;
                    for (int b2 = 0; b2 < 5; b2++)
                        window |= (uint64_t)base[b2] << (8 * b2);
                    uint32_t v = (uint32_t)((window >> shift) &
                                            ((bw < 32) ? ((1u << bw) - 1u)
                                                       : 0xFFFFFFFFu));
                    if (v >= pd.dict_n) { bad = true; break; }
                    dst[run.out_pos + i] = dict[v];
                }
                if (bad) {
                    atomicAdd(err_flag, 1ull);
                    break;
                }
            }
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Streaming parity mode (hx_scan): emit the filtered, deduplicated rows
// themselves. Appends survivors (series, ts, value) unordered; the host then
// radix-sorts by (series, ts) — MergeStream's PK order (equal PKs cannot
// survive dedup, so no stability requirement remains).
// ---------------------------------------------------------------------------
struct ScanRowsParams {
    AggParams P;              // filter/dedup context (table unused)
    uint32_t rg_first, rg_last;   // process rgs [first, last) (one segment)
    uint64_t* out_series;
    long long* out_ts;
    double* out_value;
    uint64_t* out_seq;        // per-row __seq__ of survivors (nullable)
    int32_t seq_rowidx;       // 1: out_seq = (seq << 32) | staged row index
                              // — the Append sort key: equal-PK equal-seq
                              // rows keep FILE ROW ORDER (the writer's
                              // stable sort; MergeStream stream order)
    int32_t _pad;
    unsigned long long* cursor;
    unsigned long long cap;
};

extern "C" __global__ void __launch_bounds__(256)
k_scan_rows(ScanRowsParams R) {
    const AggParams& P = R.P;
    for (uint32_t rgi = R.rg_first + blockIdx.x; rgi < R.rg_last;
         rgi += gridDim.x) {
        const RgDesc rg = P.rgs[rgi];
        const uint64_t* S = (const uint64_t*)hx_ptr(P.blob, P.dec, rg.series_off);
        const int64_t* T = (const int64_t*)hx_ptr(P.blob, P.dec, rg.ts_off);
        const double* V = (const double*)hx_ptr(P.blob, P.dec, rg.val_off);
        const SstDev sst = P.ssts[rg.sst_id];
        const uint32_t n = rg.n_rows;
        const int lane = threadIdx.x & 63;
        for (uint32_t base = 0; base < n; base += blockDim.x) {
            const uint32_t r = base + threadIdx.x;
            const bool inb = r < n;
            const int64_t t = inb ? T[r] : 0;
            const uint64_t s = inb ? S[r] : 0;
            const bool live = inb && row_alive(P, rg, sst, S, T, r, n, s, t);
            // one cursor atomic per wave (guideline 12)
            const unsigned long long mask = __ballot(live);
            if (!mask) continue;
            const int leader = __ffsll((unsigned long long)mask) - 1;
            unsigned long long wave_base = 0;
            if (lane == leader)
                wave_base = atomicAdd(R.cursor,
                                      (unsigned long long)__popcll(mask));
            wave_base = __shfl(wave_base, leader, 64);
            if (live) {
                const unsigned long long j =
                    wave_base + __popcll(mask & ((1ull << lane) - 1ull));
                if (j < R.cap) {
                    R.out_series[j] = s;
                    R.out_ts[j] = t;
                    R.out_value[j] = V[r];
                    if (R.out_seq) {
                        uint64_t q = row_seq(P, sst, rg, r);
                        R.out_seq[j] = R.seq_rowidx
                            ? (q << 32) |
                                  (uint64_t)(uint32_t)(rg.row_base + r)
                            : q;
                    }
                }
            }
        }
    }
}

// ---------------------------------------------------------------------------
// Inverted-index query kernels (rfc:86-137): PLAIN BYTE_ARRAY offset walk,
// tag-equality postings filter, sorted-set intersection / adjacent-unique.
// All HBM-bound byte work; no MFMA.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
k_ba_offsets(const uint8_t* __restrict__ blob,
             const BaPageDesc* __restrict__ pages, uint32_t n_pages,
             uint64_t* __restrict__ out, unsigned long long* err_flag) {
    // the length-prefixed layout is a serial chain per page; pages decode in
    // parallel (one thread walks one page: ~8192 values)
    const uint32_t tid = blockIdx.x * blockDim.x + threadIdx.x;
    const uint32_t n_threads = gridDim.x * blockDim.x;
    for (uint32_t pg = tid; pg < n_pages; pg += n_threads) {
        const BaPageDesc pd = pages[pg];
        uint64_t pos = 0;
        bool bad = false;
        for (uint32_t i = 0; i < pd.n_values; i++) {
            if (pos + 4 > pd.src_len) { bad = true; break; }
            const uint8_t* p = blob + pd.src_off + pos;
            uint32_t len = (uint32_t)p[0] | ((uint32_t)p[1] << 8) |
                           ((uint32_t)p[2] << 16) | ((uint32_t)p[3] << 24);
            pos += 4;
            if (len >= (1u << 20) || pos + len > pd.src_len) {
                bad = true;
                break;
            }
            out[pd.first_row + i] = ((pd.src_off + pos) << 20) | len;
            pos += len;
        }
        if (bad) atomicAdd(err_flag, 1ull);
    }
}

__device__ __forceinline__ bool bytes_eq(const uint8_t* a, const uint8_t* b,
                                         uint32_t n) {
    uint32_t i = 0;
    for (; i + 8 <= n; i += 8) {
        uint64_t x, y;
        __builtin_memcpy(&x, a + i, 8);
        __builtin_memcpy(&y, b + i, 8);
        if (x != y) return false;
    }
    for (; i < n; i++)
        if (a[i] != b[i]) return false;
    return true;
}

extern "C" __global__ void __launch_bounds__(256)
k_tag_filter(TagFilterParams F) {
    const int lane = threadIdx.x & 63;
    for (int64_t r = blockIdx.x * blockDim.x + threadIdx.x;
         __any(r < F.n_rows); r += (int64_t)blockDim.x * gridDim.x) {
        bool hit = false;
        uint64_t id = 0;
        if (r < F.n_rows) {
            const uint64_t ko = F.key_offlen[r];
            const uint32_t klen = (uint32_t)(ko & 0xFFFFFu);
            if (klen == F.pred_key_len &&
                bytes_eq(F.blob + (ko >> 20), F.pred_key, klen)) {
                const uint64_t vo = F.val_offlen[r];
                const uint32_t vlen = (uint32_t)(vo & 0xFFFFFu);
                if (vlen == F.pred_val_len &&
                    bytes_eq(F.blob + (vo >> 20), F.pred_val, vlen)) {
                    hit = true;
                    id = F.tsid[r];
                }
            }
        }
        const unsigned long long mask = __ballot(hit);
        if (!mask) continue;
        const int leader = __ffsll((unsigned long long)mask) - 1;
        unsigned long long base = 0;
        if (lane == leader)
            base = atomicAdd(F.cursor, (unsigned long long)__popcll(mask));
        base = __shfl(base, leader, 64);
        if (hit) {
            const unsigned long long j =
                base + __popcll(mask & ((1ull << lane) - 1ull));
            if (j < F.cap) F.out[j] = id;
        }
    }
}

// gather variable-length byte values: row i copies its bytes (handle =
// blob_off << 20 | len, from k_ba_offsets) to out[dst_off[i]..dst_off[i+1])
struct CopyBytesParams {
    const uint8_t* blob;
    const uint64_t* handles;    // per OUTPUT row, already permuted
    const int64_t* dst_off;     // n+1 prefix offsets into out
    uint8_t* out;
    uint32_t n;
};

extern "C" __global__ void __launch_bounds__(256)
k_copy_bytes(CopyBytesParams C) {
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < C.n;
         r += blockDim.x * gridDim.x) {
        const uint64_t h = C.handles[r];
        const uint8_t* src = C.blob + (h >> 20);
        const uint32_t len = (uint32_t)(h & 0xFFFFFu);
        uint8_t* dst = C.out + C.dst_off[r];
        for (uint32_t i = 0; i < len; i++) dst[i] = src[i];
    }
}

// sorted-set intersection: keep a[i] iff it appears in sorted b[0..n_b)
extern "C" __global__ void __launch_bounds__(256)
k_tsid_intersect(const uint64_t* __restrict__ a, unsigned long long n_a,
                 const uint64_t* __restrict__ b, unsigned long long n_b,
                 uint64_t* __restrict__ out, unsigned long long* cursor) {
    const int lane = threadIdx.x & 63;
    for (unsigned long long r = blockIdx.x * blockDim.x + threadIdx.x;
         __any(r < n_a); r += (unsigned long long)blockDim.x * gridDim.x) {
        bool hit = false;
        uint64_t v = 0;
        if (r < n_a) {
            v = a[r];
            unsigned long long lo = 0, hi = n_b;
            while (lo < hi) {
                unsigned long long mid = (lo + hi) >> 1;
                if (b[mid] < v) lo = mid + 1;
                else hi = mid;
            }
            hit = lo < n_b && b[lo] == v;
        }
        const unsigned long long mask = __ballot(hit);
        if (!mask) continue;
        const int leader = __ffsll((unsigned long long)mask) - 1;
        unsigned long long base = 0;
        if (lane == leader)
            base = atomicAdd(cursor, (unsigned long long)__popcll(mask));
        base = __shfl(base, leader, 64);
        if (hit)
            out[base + __popcll(mask & ((1ull << lane) - 1ull))] = v;
    }
}

// adjacent-unique over a SORTED array
extern "C" __global__ void __launch_bounds__(256)
k_unique_u64(const uint64_t* __restrict__ in, unsigned long long n,
             uint64_t* __restrict__ out, unsigned long long* cursor) {
    const int lane = threadIdx.x & 63;
    for (unsigned long long r = blockIdx.x * blockDim.x + threadIdx.x;
         __any(r < n); r += (unsigned long long)blockDim.x * gridDim.x) {
        const bool keep = (r < n) && (r == 0 || in[r] != in[r - 1]);
        const unsigned long long mask = __ballot(keep);
        if (!mask) continue;
        const int leader = __ffsll((unsigned long long)mask) - 1;
        unsigned long long base = 0;
        if (lane == leader)
            base = atomicAdd(cursor, (unsigned long long)__popcll(mask));
        base = __shfl(base, leader, 64);
        if (keep)
            out[base + __popcll(mask & ((1ull << lane) - 1ull))] = in[r];
    }
}

// ---------------------------------------------------------------------------
// Result compaction: live slots -> dense arrays (unsorted; host sorts with
// rocPRIM then gathers).
// ---------------------------------------------------------------------------
struct CompactParams {
    AggTable table;
    uint32_t n_slots;
    uint32_t ops;
    int32_t key_claim;
    int64_t bucket_ms;
    // direct-indexed bucket mode
    int64_t lo_bucket;
    uint32_t n_buckets;
    uint32_t bstride;
    const uint8_t* bstore;
    uint64_t* out_series;
    long long* out_bucket;
    double* out_sum;
    unsigned long long* out_cnt;
    double* out_min;
    double* out_max;
    unsigned long long* n_out;
};

extern "C" __global__ void __launch_bounds__(256)
k_compact(CompactParams C) {
    const int lane = threadIdx.x & 63;
    const uint32_t stride = blockDim.x * gridDim.x;
    if (C.n_buckets) {  // direct-indexed bucket mode: sweep (slot, bucket)
        const size_t total = (size_t)C.n_slots * C.n_buckets;
        for (size_t p = blockIdx.x * blockDim.x + threadIdx.x;
             __any(p < total); p += stride) {
            bool live = false;
            uint64_t key = 0;
            const uint8_t* brow = nullptr;
            if (p < total) {
                const uint32_t slot_i = (uint32_t)(p / C.n_buckets);
                key = *(const uint64_t*)(C.table.slab +
                                         (size_t)slot_i * C.table.stride);
                brow = C.bstore + p * C.bstride;
                live = key != KEY_EMPTY &&
                       *(const unsigned long long*)(brow + 8) > 0;
            }
            const unsigned long long mask = __ballot(live);
            if (!mask) continue;
            const int leader = __ffsll((unsigned long long)mask) - 1;
            unsigned long long wave_base = 0;
            if (lane == leader)
                wave_base =
                    atomicAdd(C.n_out, (unsigned long long)__popcll(mask));
            wave_base = __shfl(wave_base, leader, 64);
            if (live) {
                const unsigned long long j =
                    wave_base + __popcll(mask & ((1ull << lane) - 1ull));
                C.out_series[j] = key;
                C.out_bucket[j] =
                    C.lo_bucket + (long long)(p % C.n_buckets);
                if (C.out_sum) C.out_sum[j] = *(const double*)brow;
                if (C.out_cnt)
                    C.out_cnt[j] = *(const unsigned long long*)(brow + 8);
                if (C.out_min)
                    C.out_min[j] = ordered_f64(
                        *(const unsigned long long*)(brow + 16));
                if (C.out_max)
                    C.out_max[j] = ordered_f64(
                        *(const unsigned long long*)(brow + 24));
            }
        }
        return;
    }
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
         __any(i < C.n_slots); i += stride) {
        const uint8_t* slot = C.table.slab + (size_t)i * C.table.stride;
        const bool live = (i < C.n_slots) &&
                          (C.key_claim ? (*(const uint64_t*)slot != KEY_EMPTY)
                                       : (*(const uint32_t*)slot == 2u));
        const unsigned long long mask = __ballot(live);
        if (!mask) continue;
        // one atomic per wave (guideline 12): first live lane reserves
        const int leader = __ffsll((unsigned long long)mask) - 1;
        unsigned long long wave_base = 0;
        if (lane == leader)
            wave_base = atomicAdd(C.n_out, (unsigned long long)__popcll(mask));
        wave_base = __shfl(wave_base, leader, 64);
        if (live) {
            const unsigned long long j =
                wave_base + __popcll(mask & ((1ull << lane) - 1ull));
            if (C.key_claim && C.table.rep) {
                C.out_series[j] = *(const uint64_t*)slot;
                double sum = 0, mnv = HUGE_VAL, mxv = -HUGE_VAL;
                unsigned long long cntv = 0;
                for (uint32_t x = 0; x < 8; x++) {
                    const uint8_t* r2 =
                        C.table.rep +
                        ((size_t)x * (C.table.mask + 1ull) + i) *
                            C.table.rep_stride;
                    sum += *(const double*)r2;
                    cntv += *(const unsigned long long*)(r2 + 8);
                    if (C.out_min)
                        mnv = fmin(mnv, ordered_f64(
                                            *(const unsigned long long*)(r2 + 16)));
                    if (C.out_max)
                        mxv = fmax(mxv, ordered_f64(
                                            *(const unsigned long long*)(r2 + 24)));
                }
                if (C.out_sum) C.out_sum[j] = sum;
                if (C.out_cnt) C.out_cnt[j] = cntv;
                if (C.out_min) C.out_min[j] = mnv;
                if (C.out_max) C.out_max[j] = mxv;
            } else if (C.key_claim) {
                C.out_series[j] = *(const uint64_t*)slot;
                if (C.out_sum) C.out_sum[j] = *(const double*)(slot + 8);
                if (C.out_cnt)
                    C.out_cnt[j] = *(const unsigned long long*)(slot + 16);
                if (C.out_min)
                    C.out_min[j] = ordered_f64(
                        *(const unsigned long long*)(slot + 24));
                if (C.out_max)
                    C.out_max[j] = ordered_f64(
                        *(const unsigned long long*)(slot + 32));
            } else {
                C.out_series[j] = *(const uint64_t*)(slot + 8);
                if (C.bucket_ms)
                    C.out_bucket[j] = *(const long long*)(slot + 16);
                if (C.out_sum) C.out_sum[j] = *(const double*)(slot + 24);
                if (C.out_cnt)
                    C.out_cnt[j] = *(const unsigned long long*)(slot + 32);
                if (C.out_min)
                    C.out_min[j] = ordered_f64(
                        *(const unsigned long long*)(slot + 40));
                if (C.out_max)
                    C.out_max[j] = ordered_f64(
                        *(const unsigned long long*)(slot + 48));
            }
        }
    }
}

// ---- two-phase compaction (non-bucket slab/state tables) ------------------
// The single-pass k_compact reserves output positions with one agent-scope
// atomicAdd per live wave on ONE counter word. At headline fill (10M live in
// a 16M-slot table) that is ~250k serialized hot-word RMWs ≈ 2.8 ms — the
// same coherence-point storm the scan kernel's fill counter had (§9c). The
// two-phase form removes the hot word entirely: per-block live counts
// (no atomics), a one-block exclusive scan, then an atomic-free write pass
// positioned by block base + LDS-local offset. Sweep traffic is paid twice
// but is only ~0.5 GB at headline size.

__device__ __forceinline__ bool compact_live(const CompactParams& C,
                                             size_t p) {
    if (C.n_buckets) {  // direct-indexed bucket mode: p = (slot, bucket)
        const uint32_t slot_i = (uint32_t)(p / C.n_buckets);
        const uint64_t key = *(const uint64_t*)(
            C.table.slab + (size_t)slot_i * C.table.stride);
        return key != KEY_EMPTY &&
               *(const unsigned long long*)(C.bstore + p * C.bstride + 8) > 0;
    }
    const uint8_t* slot = C.table.slab + p * C.table.stride;
    return C.key_claim ? (*(const uint64_t*)slot != KEY_EMPTY)
                       : (*(const uint32_t*)slot == 2u);
}

__device__ __forceinline__ void compact_emit(const CompactParams& C,
                                             size_t p,
                                             unsigned long long j) {
    if (C.n_buckets) {
        const uint32_t slot_i = (uint32_t)(p / C.n_buckets);
        const uint8_t* brow = C.bstore + p * C.bstride;
        C.out_series[j] = *(const uint64_t*)(
            C.table.slab + (size_t)slot_i * C.table.stride);
        C.out_bucket[j] = C.lo_bucket + (long long)(p % C.n_buckets);
        if (C.out_sum) C.out_sum[j] = *(const double*)brow;
        if (C.out_cnt)
            C.out_cnt[j] = *(const unsigned long long*)(brow + 8);
        if (C.out_min)
            C.out_min[j] =
                ordered_f64(*(const unsigned long long*)(brow + 16));
        if (C.out_max)
            C.out_max[j] =
                ordered_f64(*(const unsigned long long*)(brow + 24));
        return;
    }
    const uint32_t i = (uint32_t)p;
    const uint8_t* slot = C.table.slab + (size_t)i * C.table.stride;
    if (C.key_claim && C.table.rep) {
        C.out_series[j] = *(const uint64_t*)slot;
        double sum = 0, mnv = HUGE_VAL, mxv = -HUGE_VAL;
        unsigned long long cntv = 0;
        for (uint32_t x = 0; x < 8; x++) {
            const uint8_t* r2 =
                C.table.rep +
                ((size_t)x * (C.table.mask + 1ull) + i) * C.table.rep_stride;
            sum += *(const double*)r2;
            cntv += *(const unsigned long long*)(r2 + 8);
            if (C.out_min)
                mnv = fmin(mnv,
                           ordered_f64(*(const unsigned long long*)(r2 + 16)));
            if (C.out_max)
                mxv = fmax(mxv,
                           ordered_f64(*(const unsigned long long*)(r2 + 24)));
        }
        if (C.out_sum) C.out_sum[j] = sum;
        if (C.out_cnt) C.out_cnt[j] = cntv;
        if (C.out_min) C.out_min[j] = mnv;
        if (C.out_max) C.out_max[j] = mxv;
    } else if (C.key_claim) {
        C.out_series[j] = *(const uint64_t*)slot;
        if (C.out_sum) C.out_sum[j] = *(const double*)(slot + 8);
        if (C.out_cnt) C.out_cnt[j] = *(const unsigned long long*)(slot + 16);
        if (C.out_min)
            C.out_min[j] =
                ordered_f64(*(const unsigned long long*)(slot + 24));
        if (C.out_max)
            C.out_max[j] =
                ordered_f64(*(const unsigned long long*)(slot + 32));
    } else {
        C.out_series[j] = *(const uint64_t*)(slot + 8);
        if (C.bucket_ms) C.out_bucket[j] = *(const long long*)(slot + 16);
        if (C.out_sum) C.out_sum[j] = *(const double*)(slot + 24);
        if (C.out_cnt) C.out_cnt[j] = *(const unsigned long long*)(slot + 32);
        if (C.out_min)
            C.out_min[j] =
                ordered_f64(*(const unsigned long long*)(slot + 40));
        if (C.out_max)
            C.out_max[j] =
                ordered_f64(*(const unsigned long long*)(slot + 48));
    }
}

__device__ __forceinline__ size_t compact_total(const CompactParams& C) {
    return C.n_buckets ? (size_t)C.n_slots * C.n_buckets : (size_t)C.n_slots;
}

extern "C" __global__ void __launch_bounds__(256)
k_compact_count(CompactParams C, uint32_t* __restrict__ counts) {
    __shared__ uint32_t wsum[4];
    const size_t total = compact_total(C);
    const size_t chunk = (total + gridDim.x - 1) / gridDim.x;
    const size_t beg = blockIdx.x * chunk;
    const size_t end = beg + chunk < total ? beg + chunk : total;
    uint32_t cnt = 0;
    for (size_t i0 = beg; i0 < end; i0 += blockDim.x) {
        const size_t i = i0 + threadIdx.x;
        const bool live = i < end && compact_live(C, i);
        const unsigned long long m = __ballot(live);
        if ((threadIdx.x & 63) == 0) cnt += (uint32_t)__popcll(m);
    }
    if ((threadIdx.x & 63) == 0) wsum[threadIdx.x >> 6] = cnt;
    __syncthreads();
    if (threadIdx.x == 0)
        counts[blockIdx.x] = wsum[0] + wsum[1] + wsum[2] + wsum[3];
}

// One-block exclusive scan of <=2048 per-block counts -> bases + total.
extern "C" __global__ void __launch_bounds__(256)
k_scan_counts(const uint32_t* __restrict__ counts, uint32_t nb,
              uint32_t* __restrict__ bases,
              unsigned long long* __restrict__ n_out) {
    __shared__ uint32_t tsum[256];
    const uint32_t per = (nb + 255) / 256;
    const uint32_t b0 = threadIdx.x * per;
    const uint32_t b1 = b0 + per < nb ? b0 + per : nb;
    uint32_t s = 0;
    for (uint32_t j = b0; j < b1; j++) s += counts[j];
    tsum[threadIdx.x] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint32_t acc = 0;
        for (int t = 0; t < 256; t++) {
            const uint32_t v = tsum[t];
            tsum[t] = acc;
            acc += v;
        }
        *n_out = acc;
    }
    __syncthreads();
    uint32_t acc = tsum[threadIdx.x];
    for (uint32_t j = b0; j < b1; j++) {
        bases[j] = acc;
        acc += counts[j];
    }
}

extern "C" __global__ void __launch_bounds__(256)
k_compact_write(CompactParams C, const uint32_t* __restrict__ bases) {
    __shared__ uint32_t blk_off;
    if (threadIdx.x == 0) blk_off = 0;
    __syncthreads();
    const size_t total = compact_total(C);
    const size_t chunk = (total + gridDim.x - 1) / gridDim.x;
    const size_t beg = blockIdx.x * chunk;
    const size_t end = beg + chunk < total ? beg + chunk : total;
    const unsigned long long base = bases[blockIdx.x];
    const int lane = threadIdx.x & 63;
    for (size_t i0 = beg; i0 < end; i0 += blockDim.x) {
        const size_t i = i0 + threadIdx.x;
        const bool live = i < end && compact_live(C, i);
        const unsigned long long mask = __ballot(live);
        if (!mask) continue;
        const int leader = __ffsll((unsigned long long)mask) - 1;
        uint32_t wave_base = 0;
        if (lane == leader)
            wave_base = atomicAdd(&blk_off, (uint32_t)__popcll(mask));
        wave_base = __shfl(wave_base, leader, 64);
        if (live)
            compact_emit(C, i,
                         base + wave_base +
                             __popcll(mask & ((1ull << lane) - 1ull)));
    }
}

// Gather 8-byte elements by permutation (applies the sort order).
struct GatherMulti {
    const unsigned long long* src[8];
    unsigned long long* dst;   // dst array a at dst + a*n
    const uint32_t* perm;
    uint32_t n;
    uint32_t n_arrays;
};

extern "C" __global__ void __launch_bounds__(256)
k_gather_multi(GatherMulti g) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < g.n;
         i += blockDim.x * gridDim.x) {
        const uint32_t p = g.perm[i];
        for (uint32_t a = 0; a < g.n_arrays; a++)
            g.dst[(size_t)a * g.n + i] = g.src[a][p];
    }
}

extern "C" __global__ void __launch_bounds__(256)
k_gather_u64(const unsigned long long* in, const uint32_t* perm,
             unsigned long long* out, uint32_t n) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += blockDim.x * gridDim.x)
        out[i] = in[perm[i]];
}

extern "C" __global__ void __launch_bounds__(256)
k_avg(const double* sum, const unsigned long long* cnt, double* avg, uint32_t n) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += blockDim.x * gridDim.x)
        avg[i] = sum[i] / (double)cnt[i];
}

// ---------------------------------------------------------------------------
// Dense materialization (overlap path / streaming mode): copy staged PLAIN
// page payloads into dense 8B arrays. One unit per page, grid-stride.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
k_copy_u64(const uint8_t* blob, const uint8_t* dec_in, uint8_t* dec,
           const CopyDesc* descs, uint32_t n_descs) {
    for (uint32_t d = blockIdx.x; d < n_descs; d += gridDim.x) {
        CopyDesc c = descs[d];
        const uint64_t* src = (const uint64_t*)hx_ptr(blob, dec_in, c.src_off);
        uint64_t* dst = (uint64_t*)(dec + (c.dst_off & OFF_MASK));
        for (uint32_t i = threadIdx.x; i < c.n_values; i += blockDim.x)
            dst[i] = src[i];
    }
}

// ---------------------------------------------------------------------------
// Snappy raw-block decompression (the reference's default page codec,
// config.rs:120-133). One WAVE per page: the tag stream is inherently
// sequential, so every lane parses the (wave-uniform) element headers and
// the 64 lanes copy the element's bytes cooperatively. Overlapping copies
// (offset < length) use the periodic-pattern equivalence of the sequential
// byte copy. Parallelism comes from the page count (~one 64KB page per
// 8192-row column chunk).
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint32_t snappy_varint(const uint8_t* p,
                                                  uint32_t& pos, uint32_t len,
                                                  int* err) {
    uint32_t r = 0;
    int s = 0;
    for (;;) {
        if (pos >= len || s > 28) { *err = 1; return 0; }
        uint8_t b = p[pos++];
        r |= (uint32_t)(b & 0x7f) << s;
        if (!(b & 0x80)) return r;
        s += 7;
    }
}

// Per-wave LDS mirror of the most recent output bytes: back-references with
// offset <= SNAP_MIRROR read the mirror (LDS, wave-ordered) instead of the
// global dst — the old kernel drained ALL outstanding stores
// (`s_waitcnt vmcnt(0)`, ~600-900 cycles) before EVERY match, and the
// ts/series pages of PLAIN metric data are nothing but small periodic
// matches (measured 94 ms/step at the 1B snappy shape, ~4x the aggregate
// kernel). Offsets beyond the mirror still take the drained global path.
// Snappy decode, fourth design. The decode loop is VALU-ISSUE-BOUND: a
// ts/series page is thousands of tiny matches, and each one paid ~500
// instructions of copy setup, exec-mask bookkeeping and flush checks (the
// memory-level rewrites v2/v3 moved the bytes into LDS and changed almost
// nothing). Two levers here:
//   1. FUSE runs of same-offset matches at parse time: periodic metric
//      columns compress into long chains of off=8/16 matches whose fused
//      form is ONE long periodic copy — the per-tag cost drops to the bare
//      parse (~40 instructions), the copy amortizes over the fused length;
//   2. 4 KB LDS output ring per wave (16 KB per 256-thread block => full
//      32-wave occupancy), flushed to HBM in coalesced spans; matches read
//      the ring, never global dst (no vmcnt round trips).
#define SNAP_RING 4096u

struct SnapStream {
    const uint64_t* words;   // aligned view of the page payload
    uint64_t w0, w1, w2;     // words [wi, wi+3)
    uint32_t wi;             // aligned word index of w0
    uint32_t n_words;        // ceil(payload/8) — loads are clamped

    __device__ void init(const uint8_t* src, uint32_t clen) {
        words = (const uint64_t*)src;
        n_words = (clen + 7u) >> 3;
        wi = 0;
        w0 = n_words > 0 ? words[0] : 0;
        w1 = n_words > 1 ? words[1] : 0;
        w2 = n_words > 2 ? words[2] : 0;
    }
    __device__ __forceinline__ void advance_to(uint32_t pos) {
        const uint32_t target = pos >> 3;
        if (target >= wi + 3) {  // long jump (after a big literal): reseat
            wi = target;
            w0 = wi < n_words ? words[wi] : 0;
            w1 = wi + 1 < n_words ? words[wi + 1] : 0;
            w2 = wi + 2 < n_words ? words[wi + 2] : 0;
            return;
        }
        while (target > wi) {
            wi++;
            w0 = w1;
            w1 = w2;
            w2 = (wi + 2 < n_words) ? words[wi + 2] : 0;
        }
    }
    // 8 bytes starting at byte `pos` (pos in [wi*8, wi*8+8))
    __device__ __forceinline__ uint64_t peek8(uint32_t pos) {
        const uint32_t sh = (pos & 7u) * 8u;
        return sh ? ((w0 >> sh) | (w1 << (64 - sh))) : w0;
    }
};

// flush ring bytes [f, f+len) to dst (coalesced u32 stores when everything
// is 4-aligned, else bytes). Ring spans wrap at SNAP_RING.
__device__ __forceinline__ void snap_flush(uint8_t* __restrict__ dst,
                                           const uint8_t* ring, uint32_t f,
                                           uint32_t len, uint32_t lane) {
    while (len) {
        const uint32_t r0 = f & (SNAP_RING - 1);
        const uint32_t span = min(len, SNAP_RING - r0);
        if ((r0 & 3u) == 0 && (f & 3u) == 0 && (span & 3u) == 0) {
            const uint32_t words = span >> 2;
            for (uint32_t i = lane; i < words; i += 64) {
                uint32_t v;
                __builtin_memcpy(&v, ring + r0 + 4 * i, 4);
                *(uint32_t*)(dst + f + 4 * i) = v;
            }
        } else {
            for (uint32_t i = lane; i < span; i += 64)
                dst[f + i] = ring[r0 + i];
        }
        f += span;
        len -= span;
    }
}

template <int MINW>
__global__ void __launch_bounds__(256, MINW)
k_snappy_decompress(const uint8_t* __restrict__ blob, uint8_t* __restrict__ dec,
                    const SnappyPageDesc* __restrict__ pages, uint32_t n_pages,
                    unsigned long long* err_flag) {
    __shared__ uint8_t ring_all[4][SNAP_RING];
    const uint32_t lane = threadIdx.x & 63;
    uint8_t* const ring = ring_all[(threadIdx.x >> 6) & 3];
    const uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const uint32_t n_waves = (gridDim.x * blockDim.x) >> 6;
    for (uint32_t pg = wave; pg < n_pages; pg += n_waves) {
        const SnappyPageDesc pd = pages[pg];
        const uint8_t* src = blob + (pd.src_off & OFF_MASK);
        uint8_t* dst = dec + (pd.dst_off & OFF_MASK);
        const uint32_t clen = pd.comp_len;
        int err = 0;
        SnapStream st;
        st.init(src, clen);
        uint32_t pos = 0;
        uint64_t v = st.peek8(0);
        uint32_t ulen = 0;
        {
            int sh = 0;
            for (;;) {
                if (pos >= clen || sh > 28 || pos >= 7) { err = 1; break; }
                uint8_t b = (uint8_t)(v >> (8 * pos));
                ulen |= (uint32_t)(b & 0x7f) << sh;
                pos++;
                if (!(b & 0x80)) break;
                sh += 7;
            }
        }
        if (err || ulen != pd.uncomp_len) {
            if (lane == 0) atomicAdd(err_flag, 1ull);
            continue;
        }
        uint32_t d = 0;        // output cursor
        uint32_t flushed = 0;  // dst bytes already written
        // Register sliding window (v5 fast path): series pages are 8k
        // ALTERNATING {7-byte literal, off=8 match} elements per page —
        // per-element cost must be tens of instructions, not hundreds.
        // tail0 = output[d-8, d), tail1 = output[d-16, d-8) (LE-packed);
        // pend accumulates the aligned 8-byte word at (d & ~7), written
        // with ONE ds_write_b64 per 8 output bytes. Writing the full pend
        // word early is safe: the slop bytes at [d, dA+8) sit beyond the
        // cursor in the ring and are rewritten before any flush reads them.
        uint64_t tail0 = 0, tail1 = 0, pend = 0;
        const uint64_t* const ring64 = (const uint64_t*)ring;
        // append n (1..8) bytes (LE in low bytes of nv) to the output
        auto rwin_append = [&](uint64_t nv, uint32_t n) {
            const uint32_t sh = 8u * n;
            if (n == 8) {
                tail1 = tail0;
                tail0 = nv;
            } else {
                tail1 = (tail1 >> sh) | (tail0 << (64 - sh));
                tail0 = (tail0 >> sh) | (nv << (64 - sh));
            }
            const uint32_t a = d & 7u;
            pend |= nv << (8u * a);   // a == 0 shifts by 0
            if (a + n >= 8) {
                if (lane == 0)
                    *(uint64_t*)(ring + ((d - a) & (SNAP_RING - 1) & ~7u)) =
                        pend;
                pend = a ? (nv >> (8u * (8u - a))) : 0;
            }
            d += n;
        };
        // spill the pending partial word (slop-safe) before any path that
        // reads/writes the ring directly or flushes
        auto rwin_spill = [&]() {
            if ((d & 7u) && lane == 0)
                *(uint64_t*)(ring + (d & (SNAP_RING - 1) & ~7u)) = pend;
        };
        // reload tail/pend from the ring after direct-ring paths
        auto rwin_reload = [&]() {
            const uint32_t a = d & 7u;
            const uint32_t base = (d - a) & (SNAP_RING - 1);
            const uint64_t w0 = ring64[base >> 3];
            const uint64_t w1 = ring64[((base + SNAP_RING - 8) &
                                        (SNAP_RING - 1)) >> 3];
            const uint64_t w2 = ring64[((base + SNAP_RING - 16) &
                                        (SNAP_RING - 1)) >> 3];
            if (a == 0) {
                pend = 0;
                tail0 = w1;
                tail1 = w2;
            } else {
                const uint32_t sh = 8u * a;
                pend = w0 & ((1ull << sh) - 1u);
                tail0 = (w1 >> sh) | (w0 << (64 - sh));
                tail1 = (w2 >> sh) | (w1 << (64 - sh));
            }
        };
        while (pos < clen && d < ulen && !err) {
            st.advance_to(pos);
            v = st.peek8(pos);
            uint32_t tag = (uint32_t)(v & 0xFFu);
            uint32_t kind = tag & 3u;
            // ---- v6 BULK-PAIR path: series pages are runs of the exact
            // 10-byte pair {lit7 (tag 0x18), match off=8 len=9 (tag 0x15,
            // off byte 0x08)} — 16 output bytes per pair, and the pair's
            // output is [b0..b6, p, b0..b6, p] where p = the byte at d-1,
            // INVARIANT across the run (the off-8 source telescopes).
            // 64 lanes decode 64 pairs per wave iteration: the serial
            // element loop costs ~500 instructions per element; this path
            // costs ~2.
            if (tag == 0x18u && d >= 1 && pos + 10 <= clen) {
                rwin_spill();
                const uint8_t pbyte =
                    ring[(d - 1) & (SNAP_RING - 1)];
                uint32_t consumed = 0;
                for (;;) {
                    const uint32_t pp = pos + 10u * lane;
                    bool ok = pp + 10 <= clen &&
                              d + 16u * (lane + 1u) <= ulen;
                    uint8_t b[7];
                    if (ok) {
                        ok = src[pp] == 0x18u && src[pp + 8] == 0x15u &&
                             src[pp + 9] == 0x08u;
                        if (ok)
                            for (int k = 0; k < 7; k++) b[k] = src[pp + 1 + k];
                    }
                    const unsigned long long bm = __ballot(ok);
                    const uint32_t m =
                        (~bm == 0ull) ? 64u
                                      : (uint32_t)(__ffsll((long long)~bm) - 1);
                    if (m == 0) break;
                    // ring room for 16*m bytes (+8 slop margin)
                    if (d + 16u * m + 8u > flushed + SNAP_RING) {
                        uint32_t want = d + 16u * m - (SNAP_RING / 2);
                        uint32_t take = want > flushed ? want - flushed : 0;
                        if (take > d - flushed) take = d - flushed;
                        snap_flush(dst, ring, flushed, take, lane);
                        flushed += take;
                    }
                    if (lane < m) {
                        const uint32_t base = d + 16u * lane;
                        for (int k = 0; k < 7; k++) {
                            ring[(base + k) & (SNAP_RING - 1)] = b[k];
                            ring[(base + 8 + k) & (SNAP_RING - 1)] = b[k];
                        }
                        ring[(base + 7) & (SNAP_RING - 1)] = pbyte;
                        ring[(base + 15) & (SNAP_RING - 1)] = pbyte;
                    }
                    d += 16u * m;
                    pos += 10u * m;
                    consumed += m;
                    if (m < 64u) break;
                }
                if (consumed) {
                    rwin_reload();
                    continue;
                }
                // zero pairs matched: fall through to the per-element paths
                // (the register window is still consistent after the spill)
            }
            // ---- v5 fast path: tiny literal / pow2-offset short match ----
            if (kind == 0) {
                // len <= 7: the literal's bytes are exactly the upper 7
                // bytes of the peek window (tag occupies byte 0)
                const uint32_t len = (tag >> 2) + 1;
                if (len <= 7 && pos + 1 + len <= clen && d + len <= ulen &&
                    d + len + 8 <= flushed + SNAP_RING) {
                    const uint64_t nv = (v >> 8) & ((1ull << (8 * len)) - 1);
                    rwin_append(nv, len);
                    pos += 1 + len;
                    continue;
                }
            } else if (kind == 1 || kind == 2) {
                uint32_t len, off, hdr;
                if (kind == 1) {
                    len = ((tag >> 2) & 0x7u) + 4;
                    off = (uint32_t)((tag >> 5) << 8) |
                          (uint32_t)((v >> 8) & 0xFFu);
                    hdr = 2;
                } else {
                    len = (tag >> 2) + 1;
                    off = (uint32_t)((v >> 8) & 0xFFFFu);
                    hdr = 3;
                }
                if ((off == 1 || off == 2 || off == 4 || off == 8) &&
                    off <= d && pos + hdr <= clen && d + len <= ulen &&
                    d + len + 8 <= flushed + SNAP_RING) {
                    // pow2 period divides 8: the replicated pattern word is
                    // constant across every emitted 8-byte chunk
                    uint64_t rep = tail0 >> (64 - 8 * off);
                    if (off < 8) rep |= rep << (8 * off);
                    if (off < 4) rep |= rep << 16;
                    if (off < 2) rep |= rep << 32;
                    uint32_t rem = len;
                    while (rem >= 8) {
                        rwin_append(rep, 8);
                        rem -= 8;
                    }
                    if (rem)
                        rwin_append(rep & ((1ull << (8 * rem)) - 1), rem);
                    pos += hdr;
                    continue;
                }
            }
            // ---- general paths (ring-direct): sync the register window --
            rwin_spill();
            bool slow = true;
            if (kind == 0) {  // literal
                uint32_t len = (tag >> 2) + 1;
                uint32_t hdr = 1;
                if (len > 60) {
                    const uint32_t nb = len - 60;
                    if (pos + 1 + nb > clen) { err = 1; break; }
                    len = (uint32_t)((v >> 8) &
                                     ((nb >= 4) ? 0xFFFFFFFFull
                                                : ((1ull << (8 * nb)) - 1)));
                    len += 1;
                    hdr = 1 + nb;
                }
                pos += hdr;
                if (pos + len > clen || d + len > ulen) { err = 1; break; }
                if (len >= SNAP_RING) {
                    // huge literal (incompressible value pages): flush the
                    // ring, copy src->dst directly, re-prime the ring tail
                    snap_flush(dst, ring, flushed, d - flushed, lane);
                    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
                    if ((d & 7u) == 0) {
                        // dst 8-aligned (whole-page literals start at 0):
                        // copy 8 output bytes per lane from two aligned
                        // source words + funnel shift — 8x fewer
                        // instructions than the byte loop
                        const uint64_t* sw =
                            (const uint64_t*)(src + ((pos) & ~7u));
                        const uint32_t sh = 8u * (pos & 7u);
                        // sh != 0 reads sw[w+1]: stop a word early so the
                        // last aligned read never crosses the blob slack
                        const uint32_t words =
                            sh ? ((len >> 3) ? (len >> 3) - 1 : 0)
                               : (len >> 3);
                        for (uint32_t w = lane; w < words; w += 64) {
                            uint64_t lo = sw[w];
                            uint64_t x;
                            if (sh) {
                                uint64_t hi2 = sw[w + 1];
                                x = (lo >> sh) | (hi2 << (64 - sh));
                            } else {
                                x = lo;
                            }
                            *(uint64_t*)(dst + d + 8 * w) = x;
                        }
                        for (uint32_t i = (words << 3) + lane; i < len;
                             i += 64)
                            dst[d + i] = src[pos + i];
                    } else {
                        for (uint32_t i = lane; i < len; i += 64)
                            dst[d + i] = src[pos + i];
                    }
                    for (uint32_t i = lane; i < SNAP_RING; i += 64)
                        ring[(d + len - SNAP_RING + i) & (SNAP_RING - 1)] =
                            src[pos + len - SNAP_RING + i];
                    d += len;
                    flushed = d;
                } else {
                    if (d + len + 8 > flushed + SNAP_RING) {
                        uint32_t want = d + len - (SNAP_RING / 2);
                        uint32_t take = want > flushed ? want - flushed : 0;
                        if (take > d - flushed) take = d - flushed;
                        snap_flush(dst, ring, flushed, take, lane);
                        flushed += take;
                    }
                    for (uint32_t i = lane; i < len; i += 64)
                        ring[(d + i) & (SNAP_RING - 1)] = src[pos + i];
                    d += len;
                }
                pos += len;
            } else {
                uint32_t len, off;
                if (kind == 1) {
                    len = ((tag >> 2) & 0x7u) + 4;
                    off = (uint32_t)((tag >> 5) << 8) |
                          (uint32_t)((v >> 8) & 0xFFu);
                    pos += 2;
                } else if (kind == 2) {
                    len = (tag >> 2) + 1;
                    off = (uint32_t)((v >> 8) & 0xFFFFu);
                    pos += 3;
                } else {
                    len = (tag >> 2) + 1;
                    off = (uint32_t)((v >> 8) & 0xFFFFFFFFull);
                    pos += 5;
                }
                if (pos > clen || off == 0 || off > d) { err = 1; break; }
                // FUSE adjacent same-offset matches: copy k covers
                // [d+L, d+L+len_k) from [d+L-off, ...) — the continuation
                // of one periodic/shifted copy. The fused run executes as
                // a single cooperative loop; per-tag cost = the parse.
                for (;;) {
                    st.advance_to(pos);
                    const uint64_t nv = st.peek8(pos);
                    const uint32_t ntag = (uint32_t)(nv & 0xFFu);
                    const uint32_t nkind = ntag & 3u;
                    uint32_t nlen, noff, nhdr;
                    if (nkind == 1) {
                        nlen = ((ntag >> 2) & 0x7u) + 4;
                        noff = (uint32_t)((ntag >> 5) << 8) |
                               (uint32_t)((nv >> 8) & 0xFFu);
                        nhdr = 2;
                    } else if (nkind == 2) {
                        nlen = (ntag >> 2) + 1;
                        noff = (uint32_t)((nv >> 8) & 0xFFFFu);
                        nhdr = 3;
                    } else {
                        break;   // literal or 4-byte-offset copy: stop
                    }
                    if (noff != off || pos + nhdr > clen ||
                        d + len + nlen > ulen)
                        break;
                    len += nlen;
                    pos += nhdr;
                }
                if (d + len > ulen) { err = 1; break; }
                if (off > SNAP_RING / 2) {
                    // far back-reference (not produced for this data
                    // shape): flush + drain once, then chunked global
                    // reads of the (flushed, stable) pre-run bytes
                    snap_flush(dst, ring, flushed, d - flushed, lane);
                    flushed = d;
                    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
                    const uint32_t D = d;
                    uint32_t done = 0;
                    while (done < len) {
                        const uint32_t chunk =
                            min(len - done, SNAP_RING / 2);
                        if (d + chunk + 8 > flushed + SNAP_RING) {
                            uint32_t want = d + chunk - (SNAP_RING / 2);
                            uint32_t take =
                                want > flushed ? want - flushed : 0;
                            if (take > d - flushed) take = d - flushed;
                            snap_flush(dst, ring, flushed, take, lane);
                            flushed += take;
                        }
                        for (uint32_t i = lane; i < chunk; i += 64) {
                            const uint32_t j = done + i;
                            const uint8_t b =
                                (off >= len) ? dst[D - off + j]
                                             : dst[D - off + (j % off)];
                            ring[(d + i) & (SNAP_RING - 1)] = b;
                        }
                        d += chunk;
                        done += chunk;
                    }
                } else {
                    // ring-resident source, period-doubling: after `done`
                    // bytes of the run, everything in [d - done - off, d)
                    // repeats with period (done + off), so each chunk can
                    // copy with shift P = min(done + off, 2048) — sources
                    // are all pre-chunk and at most ~2 KB behind the
                    // cursor (always ring-resident). Chunks grow
                    // geometrically: a 64 KB off=8 run takes ~40 rounds.
                    uint32_t done = 0;
                    while (done < len) {
                        // shift P must stay a MULTIPLE of off (periodicity)
                        // and <= RING/2 (so P + chunk <= RING: this chunk's
                        // writes never clobber its sources). done + off is
                        // a multiple of off by construction (done sums
                        // previous P's).
                        uint32_t P = done + off;
                        if (P > SNAP_RING / 2)
                            P = off * ((SNAP_RING / 2) / off);
                        const uint32_t chunk = min(len - done, P);
                        if (d + chunk + 8 > flushed + SNAP_RING) {
                            uint32_t want = d + chunk - (SNAP_RING / 2);
                            uint32_t take =
                                want > flushed ? want - flushed : 0;
                            if (take > d - flushed) take = d - flushed;
                            snap_flush(dst, ring, flushed, take, lane);
                            flushed += take;
                        }
                        for (uint32_t i = lane; i < chunk; i += 64)
                            ring[(d + i) & (SNAP_RING - 1)] =
                                ring[(d - P + i) & (SNAP_RING - 1)];
                        d += chunk;
                        done += chunk;
                    }
                }
            }
            if (slow) rwin_reload();
        }
        rwin_spill();
        if (!err && d == ulen) {
            snap_flush(dst, ring, flushed, d - flushed, lane);
        } else if (lane == 0) {
            atomicAdd(err_flag, 1ull);
        }
    }
}

// ---------------------------------------------------------------------------
// DELTA_BINARY_PACKED i64 decode (parquet-format Encodings.md; the encoding
// the reference's config enables for ts, config.rs:54-75). One workgroup per
// page: lane-serial header walk, parallel miniblock bit-unpack, hierarchical
// prefix sum in LDS, dense i64 output. Page <= 8192 values (the writer
// contract: row group 8192, one data page per chunk — engine validates).
// ---------------------------------------------------------------------------
#define DELTA_MAX_VALUES 8192
#define DELTA_MAX_BLOCKS 128   // >= max_values / min_block_size(128)

struct DeltaHdr {
    int32_t values_per_mb;
    int32_t n_mb_per_block;
    int32_t total_count;
    int64_t first_value;
    int32_t n_blocks;
    int32_t err;
};

__device__ inline uint64_t d_varint(const uint8_t* p, uint32_t len, uint32_t& pos,
                                    int32_t* err) {
    uint64_t r = 0;
    int s = 0;
    for (;;) {
        if (pos >= len || s > 63) { *err = 1; return 0; }
        uint8_t b = p[pos++];
        r |= (uint64_t)(b & 0x7f) << s;
        if (!(b & 0x80)) return r;
        s += 7;
    }
}
__device__ inline int64_t d_zigzag(const uint8_t* p, uint32_t len, uint32_t& pos,
                                   int32_t* err) {
    uint64_t u = d_varint(p, len, pos, err);
    return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
}

extern "C" __global__ void __launch_bounds__(256)
k_decode_delta_i64(const uint8_t* blob, uint8_t* dec,
                   const DeltaPageDesc* pages, uint32_t n_pages,
                   unsigned long long* err_flag) {
    __shared__ DeltaHdr H;
    __shared__ int64_t vals[DELTA_MAX_VALUES];          // deltas -> prefix
    __shared__ int64_t blk_min_delta[DELTA_MAX_BLOCKS];
    __shared__ uint32_t mb_off[DELTA_MAX_BLOCKS * 8];   // per-miniblock data offset
    __shared__ uint8_t mb_w[DELTA_MAX_BLOCKS * 8];
    __shared__ int64_t tsum[256 / 64];                  // per-wave partials
    __shared__ int64_t chunk_sum[256];

    for (uint32_t pg = blockIdx.x; pg < n_pages; pg += gridDim.x) {
        const DeltaPageDesc pd = pages[pg];
        const uint8_t* src = ((pd.src_off & OFF_DEC) ? dec : blob) +
                             (pd.src_off & OFF_MASK);
        const uint32_t len = pd.src_len;

        if (threadIdx.x == 0) {
            int32_t err = 0;
            uint32_t pos = 0;
            int32_t block_size = (int32_t)d_varint(src, len, pos, &err);
            int32_t n_mb = (int32_t)d_varint(src, len, pos, &err);
            int32_t total = (int32_t)d_varint(src, len, pos, &err);
            int64_t first = d_zigzag(src, len, pos, &err);
            H.err = err;
            if (!err && (n_mb <= 0 || block_size <= 0 || n_mb > 8 ||
                         block_size % n_mb != 0 || total > DELTA_MAX_VALUES ||
                         (uint32_t)total != pd.n_values ||
                         (block_size / n_mb) % 32 != 0)) {
                H.err = 2;
            }
            if (!H.err) {
                H.values_per_mb = block_size / n_mb;
                H.n_mb_per_block = n_mb;
                H.total_count = total;
                H.first_value = first;
                int32_t deltas = total - 1;
                int32_t nblk = deltas <= 0 ? 0 : (deltas + block_size - 1) / block_size;
                H.n_blocks = nblk;
                // serial walk of block headers (varints force it; 64 blocks max)
                int32_t remaining = deltas;
                for (int32_t b = 0; b < nblk && !H.err; b++) {
                    blk_min_delta[b] = d_zigzag(src, len, pos, &err);
                    if (err) { H.err = 3; break; }
                    uint32_t wpos = pos;
                    pos += n_mb;  // bit width bytes
                    if (pos > len) { H.err = 3; break; }
                    for (int32_t m = 0; m < n_mb; m++) {
                        uint8_t w = src[wpos + m];
                        mb_w[b * 8 + m] = w;
                        mb_off[b * 8 + m] = pos;
                        // data present for miniblocks that hold any values
                        if (remaining > 0) pos += (uint32_t)H.values_per_mb * w / 8;
                        remaining -= H.values_per_mb;
                    }
                    if (pos > len) { H.err = 3; }
                }
            }
        }
        __syncthreads();
        if (H.err) {
            if (threadIdx.x == 0) atomicAdd(err_flag, 1ull);
            __syncthreads();
            continue;
        }
        const int32_t total = H.total_count;
        const int32_t vpm = H.values_per_mb;
        const int32_t deltas = total - 1;

        // parallel unpack: delta j (0-based, value index j+1)
        for (int32_t j = threadIdx.x; j < deltas; j += blockDim.x) {
            int32_t blk = j / (vpm * H.n_mb_per_block);
            int32_t inblk = j - blk * vpm * H.n_mb_per_block;
            int32_t mb = inblk / vpm;
            int32_t inmb = inblk - mb * vpm;
            uint8_t w = mb_w[blk * 8 + mb];
            uint64_t raw = 0;
            if (w > 0) {
                uint64_t bitpos = (uint64_t)inmb * w;
                const uint8_t* base = src + mb_off[blk * 8 + mb] + (bitpos >> 3);
                int shift = (int)(bitpos & 7);
                // assemble enough bytes for w bits + shift (<= 9 bytes)
                uint64_t lo = 0;
                for (int k = 0; k < 8; k++) lo |= (uint64_t)base[k] << (8 * k);
                raw = lo >> shift;
                if (w + shift > 64) {
                    uint64_t hi = base[8];
                    raw |= hi << (64 - shift);
                }
                if (w < 64) raw &= ((1ull << w) - 1);
            }
            vals[j + 1] = (int64_t)raw + blk_min_delta[blk];
        }
        if (threadIdx.x == 0) vals[0] = 0;
        __syncthreads();

        // hierarchical inclusive prefix sum over vals[0..total)
        // each thread scans a contiguous chunk of 32, then chunk offsets
        const int32_t CH = 32;
        int32_t c0 = threadIdx.x * CH;
        int64_t acc = 0;
        for (int32_t j = c0; j < min(c0 + CH, total); j++) {
            acc += vals[j];
            vals[j] = acc;
        }
        chunk_sum[threadIdx.x] = acc;
        __syncthreads();
        if (threadIdx.x < 64) {  // single wave scans the 256 chunk sums
            int64_t s0 = 0;
            for (int k = 0; k < 4; k++) {
                int idx = threadIdx.x * 4 + k;
                s0 += chunk_sum[idx];
            }
            // wave inclusive scan of s0 over 64 lanes
            int64_t sc = s0;
            for (int off = 1; off < 64; off <<= 1) {
                int64_t up = __shfl_up(sc, off, 64);
                if ((int)(threadIdx.x) >= off) sc += up;
            }
            tsum[0] = 0;  // unused; keep LDS referenced
            // exclusive base for each 4-chunk group of this lane
            int64_t base4 = sc - s0;
            int64_t run = base4;
            for (int k = 0; k < 4; k++) {
                int idx = threadIdx.x * 4 + k;
                int64_t cs = chunk_sum[idx];
                chunk_sum[idx] = run;  // exclusive chunk base
                run += cs;
            }
        }
        __syncthreads();
        int64_t base = chunk_sum[threadIdx.x] + H.first_value;
        for (int32_t j = c0; j < min(c0 + CH, total); j++) vals[j] += base;
        // note: vals[0] = 0 + base = first_value ✓
        __syncthreads();

        int64_t* dst = (int64_t*)(dec + (pd.dst_off & OFF_MASK));
        for (int32_t j = threadIdx.x; j < total; j += blockDim.x)
            dst[j] = vals[j];
        __syncthreads();
    }
}

}  // namespace hx

// ---------------------------------------------------------------------------
// host-side launchers (hx_kernels.h)
// ---------------------------------------------------------------------------
#include "hx_kernels.h"
#include <algorithm>
#include <cstring>
#include <cstdlib>
#include <rocprim/device/device_radix_sort.hpp>

namespace hx {

extern "C" __global__ void __launch_bounds__(256)
k_seg_keys(const long long* ts, long long seg_ms, unsigned long long* keys,
           uint32_t n) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += blockDim.x * gridDim.x) {
        long long t = ts[i];
        long long q = t / seg_ms;
        if ((t % seg_ms) != 0 && t < 0) q--;
        keys[i] = (unsigned long long)q ^ 0x8000000000000000ull;
    }
}

extern "C" __global__ void __launch_bounds__(256)
k_iota(uint32_t* out, uint32_t n) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += blockDim.x * gridDim.x)
        out[i] = i;
}

static uint32_t grid_for(uint32_t work, uint32_t per_block) {
    uint32_t g = (work + per_block - 1) / per_block;
    return g == 0 ? 1 : (g > 2048u ? 2048u : g);
}

hipError_t launch_decode_delta(hipStream_t s, const uint8_t* blob, uint8_t* dec,
                               const DeltaPageDesc* pages, uint32_t n_pages,
                               unsigned long long* err_flag) {
    uint32_t grid = n_pages > 4096 ? 4096 : n_pages;
    hipLaunchKernelGGL(k_decode_delta_i64, dim3(grid), dim3(256), 0, s,
                       blob, dec, pages, n_pages, err_flag);
    return hipGetLastError();
}

hipError_t launch_rledict(hipStream_t s, const uint8_t* blob, uint8_t* dec,
                          const RleDictPageDesc* pages, uint32_t n_pages,
                          unsigned long long* err_flag) {
    uint32_t grid = n_pages > 4096 ? 4096 : (n_pages ? n_pages : 1);
    hipLaunchKernelGGL(k_decode_rledict, dim3(grid), dim3(256), 0, s,
                       blob, dec, pages, n_pages, err_flag);
    return hipGetLastError();
}

hipError_t launch_snappy(hipStream_t s, const uint8_t* blob, uint8_t* dec,
                         const SnappyPageDesc* pages, uint32_t n_pages,
                         unsigned long long* err_flag) {
    uint32_t waves_needed = n_pages;
    uint32_t blocks = (waves_needed + 3) / 4;  // 4 waves per 256-thread block
    if (blocks > 8192) blocks = 8192;
    if (blocks == 0) blocks = 1;
    // decode is 31% issue-busy / 62% parked at the unconstrained 111 VGPRs
    // (16 waves/CU) — unlike the aggregate it benefits from occupancy;
    // HX_SNAPPY_MINW selects. Default 6 (24 waves/CU, 440 B scratch):
    // measured best in two same-box headline A/Bs (26.1 vs 27.0/27.1 ms
    // per step against MINW 1/5 at pipeline 3; 30.4 vs 31.0/31.3 solo) —
    // the extra waves buy more than the spill costs.
    int minw = 6;
    if (const char* e = getenv("HX_SNAPPY_MINW")) minw = atoi(e);
    if (minw >= 8)
        hipLaunchKernelGGL(k_snappy_decompress<8>, dim3(blocks), dim3(256),
                           0, s, blob, dec, pages, n_pages, err_flag);
    else if (minw == 7)
        hipLaunchKernelGGL(k_snappy_decompress<7>, dim3(blocks), dim3(256),
                           0, s, blob, dec, pages, n_pages, err_flag);
    else if (minw == 6)
        hipLaunchKernelGGL(k_snappy_decompress<6>, dim3(blocks), dim3(256),
                           0, s, blob, dec, pages, n_pages, err_flag);
    else if (minw == 5)
        hipLaunchKernelGGL(k_snappy_decompress<5>, dim3(blocks), dim3(256),
                           0, s, blob, dec, pages, n_pages, err_flag);
    else
        hipLaunchKernelGGL(k_snappy_decompress<1>, dim3(blocks), dim3(256),
                           0, s, blob, dec, pages, n_pages, err_flag);
    return hipGetLastError();
}

hipError_t launch_copy_u64(hipStream_t s, const uint8_t* blob, uint8_t* dec,
                           const CopyDesc* descs, uint32_t n_descs) {
    uint32_t grid = n_descs > 4096 ? 4096 : n_descs;
    hipLaunchKernelGGL(k_copy_u64, dim3(grid), dim3(256), 0, s,
                       blob, dec, dec, descs, n_descs);
    return hipGetLastError();
}

hipError_t launch_scan_agg(hipStream_t s, const AggParams& p, uint32_t grid) {
    if (grid == 0) grid = p.n_rgs > 65535 ? 65535 : (p.n_rgs ? p.n_rgs : 1);
    hipLaunchKernelGGL(k_scan_agg, dim3(grid), dim3(256), 0, s, p);
    return hipGetLastError();
}

hipError_t launch_scan_agg_gang(hipStream_t s, const AggParams& p,
                                uint32_t gang_size, bool minmax,
                                GangParams* h_params, GangParams* d_params) {
    GangParams& G = *h_params;
    G.P = p;
    G.gang_size = gang_size;
    G.n_gangs = (p.n_rgs + gang_size - 1) / gang_size;
    G.has_mm = minmax ? 1u : 0u;
    // LDS table holds the distinct keys of one aligned unit window (host
    // slices units to <= ne/2 distinct); 4096 x 20B = 80 KiB => 2 blocks
    // (32 waves) per CU for latency hiding.
    G.ne = minmax ? 2048u : 4096u;
    if (const char* nee = getenv("HX_NE")) {
        uint32_t ne = (uint32_t)strtoul(nee, nullptr, 10);
        if (ne >= 1024 && ne <= 8192 && !(ne & (ne - 1))) G.ne = ne;
    }
    size_t lds = 320 + (size_t)G.ne * (minmax ? 36 : 20) +
                 (size_t)G.gang_size * sizeof(RgDesc);
    uint32_t grid = G.n_gangs > 4096 ? 4096 : (G.n_gangs ? G.n_gangs : 1);
    // >64 KiB dynamic LDS needs an explicit opt-in per kernel
    static bool lds_opted[2] = {false, false};
    if (!lds_opted[minmax ? 1 : 0]) {
        const void* f = minmax
            ? reinterpret_cast<const void*>(&k_scan_agg_gang<true>)
            : reinterpret_cast<const void*>(&k_scan_agg_gang<false>);
        hipError_t ae = hipFuncSetAttribute(
            f, hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        if (ae != hipSuccess) return ae;
        lds_opted[minmax ? 1 : 0] = true;
    }
    hipError_t e = hipMemcpyAsync(d_params, h_params, sizeof(GangParams),
                                  hipMemcpyHostToDevice, s);
    if (e != hipSuccess) return e;
    if (minmax)
        hipLaunchKernelGGL(k_scan_agg_gang<true>, dim3(grid), dim3(1024), lds,
                           s, d_params, p.blob, p.dec, p.rgs);
    else
        hipLaunchKernelGGL(k_scan_agg_gang<false>, dim3(grid), dim3(1024), lds,
                           s, d_params, p.blob, p.dec, p.rgs);
    return hipGetLastError();
}

hipError_t launch_sample_series(hipStream_t s, const RgDesc* rgs,
                                uint32_t n_rgs, const uint8_t* blob,
                                const uint8_t* dec, uint64_t* out) {
    uint32_t blocks = (n_rgs * 64 + 255) / 256;
    if (blocks > 65535) blocks = 65535;
    if (blocks == 0) blocks = 1;
    hipLaunchKernelGGL(k_sample_series, dim3(blocks), dim3(256), 0, s, rgs,
                       n_rgs, blob, dec, out);
    return hipGetLastError();
}

hipError_t launch_range_bounds(hipStream_t s, const AggParams& p,
                               const RangeAux& r, uint64_t* out) {
    const uint64_t total_waves = (uint64_t)(r.n_blocks + 1) * r.n_ssts;
    uint64_t blocks = (total_waves + 3) / 4;   // 4 waves per 256-thread block
    if (blocks > 262144) blocks = 262144;
    if (blocks == 0) blocks = 1;
    hipLaunchKernelGGL(k_range_bounds, dim3((uint32_t)blocks), dim3(256), 0,
                       s, p, r, out);
    return hipGetLastError();
}

hipError_t launch_scan_agg_range2(hipStream_t s, const AggParams& p,
                                  const RangeAux& r, bool minmax) {
    const size_t lds = (size_t)r.ne * (minmax ? 36 : 20);
    const void* f = minmax
                        ? reinterpret_cast<const void*>(&k_scan_agg_range2<true>)
                        : reinterpret_cast<const void*>(
                              &k_scan_agg_range2<false>);
    if (lds > 64 * 1024) {
        hipError_t e = hipFuncSetAttribute(
            f, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
        if (e != hipSuccess) return e;
    }
    if (minmax)
        hipLaunchKernelGGL(k_scan_agg_range2<true>, dim3(r.n_blocks),
                           dim3(256), lds, s, p, r);
    else
        hipLaunchKernelGGL(k_scan_agg_range2<false>, dim3(r.n_blocks),
                           dim3(256), lds, s, p, r);
    return hipGetLastError();
}

hipError_t launch_scan_agg_range(hipStream_t s, const AggParams& p,
                                 const RangeAux& r, bool minmax) {
    const size_t lds = (size_t)r.ne * (minmax ? 36 : 20);
    // occupancy: the unconstrained build allocates 118 VGPRs => 4 waves/
    // SIMD (16/CU). MINW 5/6 force the register budget down (<=96/<=80)
    // at 80/128 B of spill; after the fill-counter fix all three measure
    // within ~3% (7.1-7.4 ms) — default 1 (no spill) keeps the PMC
    // traffic at the algorithmic 1.1-1.3x instead of +7 GB of scratch.
    int minw = 1;
    if (const char* e = getenv("HX_MINW")) minw = atoi(e);
    const void* f;
    if (minmax)
        f = minw >= 6 ? (const void*)&k_scan_agg_range<true, 6>
            : minw == 5 ? (const void*)&k_scan_agg_range<true, 5>
                        : (const void*)&k_scan_agg_range<true, 1>;
    else
        f = minw >= 6 ? (const void*)&k_scan_agg_range<false, 6>
            : minw == 5 ? (const void*)&k_scan_agg_range<false, 5>
                        : (const void*)&k_scan_agg_range<false, 1>;
    // >64 KiB dynamic LDS requires the opt-in or the launch FAILS silently
    // (measured the hard way on the gang kernel). Request only what this
    // launch needs: the kernel's static shared (s_abort) counts against
    // the 160 KiB block limit, so asking for the full 160 KiB is itself
    // an invalid argument.
    if (lds > 64 * 1024) {
        hipError_t e = hipFuncSetAttribute(
            f, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
        if (e != hipSuccess) return e;
    }
    hipLaunchKernelGGL(
        reinterpret_cast<void (*)(AggParams, RangeAux)>(
            const_cast<void*>(f)),
        dim3(r.n_blocks), dim3(256), lds, s, p, r);
    return hipGetLastError();
}

hipError_t launch_scan_rows(hipStream_t s, const AggParams& p,
                            uint32_t rg_first, uint32_t rg_last,
                            uint64_t* out_series, long long* out_ts,
                            double* out_value, unsigned long long* cursor,
                            unsigned long long cap, uint64_t* out_seq,
                            int32_t seq_rowidx) {
    ScanRowsParams R;
    R.P = p;
    R.rg_first = rg_first;
    R.rg_last = rg_last;
    R.out_series = out_series;
    R.out_ts = out_ts;
    R.out_value = out_value;
    R.out_seq = out_seq;
    R.seq_rowidx = seq_rowidx;
    R._pad = 0;
    R.cursor = cursor;
    R.cap = cap;
    uint32_t n = rg_last - rg_first;
    hipLaunchKernelGGL(k_scan_rows, dim3(n > 4096 ? 4096 : (n ? n : 1)),
                       dim3(256), 0, s, R);
    return hipGetLastError();
}

hipError_t launch_gather_multi(hipStream_t s,
                               const unsigned long long* const* srcs,
                               uint32_t n_arrays, const uint32_t* perm,
                               unsigned long long* dst, uint32_t n) {
    GatherMulti g{};
    for (uint32_t a = 0; a < n_arrays && a < 8; a++) g.src[a] = srcs[a];
    g.dst = dst;
    g.perm = perm;
    g.n = n;
    g.n_arrays = n_arrays;
    hipLaunchKernelGGL(k_gather_multi, dim3(grid_for(n, 256)), dim3(256), 0, s, g);
    return hipGetLastError();
}

hipError_t launch_compact(hipStream_t s, const AggTable& t, uint32_t n_slots,
                          uint32_t ops, int32_t key_claim, int64_t bucket_ms,
                          int64_t lo_bucket, uint32_t n_buckets,
                          uint32_t bstride, const uint8_t* bstore,
                          const CompactOut& o, uint32_t* scratch) {
    CompactParams C;
    C.table = t;
    C.n_slots = n_slots;
    C.ops = ops;
    C.key_claim = key_claim;
    C.bucket_ms = bucket_ms;
    C.lo_bucket = lo_bucket;
    C.n_buckets = n_buckets;
    C.bstride = bstride;
    C.bstore = bstore;
    C.out_series = o.series;
    C.out_bucket = o.bucket;
    C.out_sum = o.sum;
    C.out_cnt = o.cnt;
    C.out_min = o.vmin;
    C.out_max = o.vmax;
    C.n_out = o.n_out;
    const size_t total =
        n_buckets ? (size_t)n_slots * n_buckets : (size_t)n_slots;
    const uint32_t grid =
        total > (size_t)2048 * 256
            ? 2048u
            : (uint32_t)((total + 255) / 256 ? (total + 255) / 256 : 1);
    // two-phase compaction (count -> scan -> write) avoids the single hot
    // n_out counter (~250k serialized agent-scope RMWs = ~3 ms at headline
    // fill). scratch = 2*grid u32 (counts, bases). NON-BUCKET ONLY: the
    // bucket sweep walks slots x n_buckets positions at low live fraction,
    // and doubling that sweep measured ~5% SLOWER than the single-pass
    // kernel's atomics (201.8 vs 192.9 ms/step same-box) — the storm is
    // proportional to LIVE waves, which bucket sweeps have few of.
    static const bool legacy = [] {
        const char* e = getenv("HX_COMPACT_LEGACY");
        return e && atoi(e) != 0;
    }();
    if (n_buckets == 0 && scratch && !legacy) {
        uint32_t* counts = scratch;
        uint32_t* bases = scratch + grid;
        hipLaunchKernelGGL(k_compact_count, dim3(grid), dim3(256), 0, s,
                           C, counts);
        hipLaunchKernelGGL(k_scan_counts, dim3(1), dim3(256), 0, s,
                           counts, grid, bases, C.n_out);
        hipLaunchKernelGGL(k_compact_write, dim3(grid), dim3(256), 0, s,
                           C, bases);
        return hipGetLastError();
    }
    hipLaunchKernelGGL(k_compact, dim3(grid), dim3(256), 0, s, C);
    return hipGetLastError();
}

hipError_t launch_gather_u64(hipStream_t s, const unsigned long long* in,
                             const uint32_t* perm, unsigned long long* out,
                             uint32_t n) {
    hipLaunchKernelGGL(k_gather_u64, dim3(grid_for(n, 256)), dim3(256), 0, s,
                       in, perm, out, n);
    return hipGetLastError();
}

hipError_t launch_avg(hipStream_t s, const double* sum,
                      const unsigned long long* cnt, double* avg, uint32_t n) {
    hipLaunchKernelGGL(k_avg, dim3(grid_for(n, 256)), dim3(256), 0, s,
                       sum, cnt, avg, n);
    return hipGetLastError();
}

extern "C" __global__ void __launch_bounds__(256)
k_xor_sign(unsigned long long* buf, uint32_t n) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += blockDim.x * gridDim.x)
        buf[i] ^= 0x8000000000000000ull;
}

hipError_t launch_xor_sign(hipStream_t s, unsigned long long* buf, uint32_t n) {
    hipLaunchKernelGGL(k_xor_sign, dim3(grid_for(n, 256)), dim3(256), 0, s, buf, n);
    return hipGetLastError();
}

hipError_t launch_seg_keys(hipStream_t s, const long long* ts, long long seg_ms,
                           unsigned long long* keys, uint32_t n) {
    hipLaunchKernelGGL(k_seg_keys, dim3(grid_for(n, 256)), dim3(256), 0, s,
                       ts, seg_ms, keys, n);
    return hipGetLastError();
}

hipError_t launch_init_slab(hipStream_t s, uint8_t* slab, uint32_t n_slots,
                            uint32_t stride) {
    hipLaunchKernelGGL(k_init_slab, dim3(grid_for(n_slots, 256)), dim3(256),
                       0, s, slab, n_slots, stride);
    return hipGetLastError();
}

hipError_t launch_init_state_slab(hipStream_t s, uint8_t* slab,
                                  uint32_t n_slots, uint32_t stride, bool mm) {
    hipLaunchKernelGGL(k_init_state_slab, dim3(grid_for(n_slots, 256)),
                       dim3(256), 0, s, slab, n_slots, stride, mm ? 1u : 0u);
    return hipGetLastError();
}

hipError_t launch_init_rep(hipStream_t s, uint8_t* rep, size_t total_slots,
                           uint32_t stride, bool mm) {
    hipLaunchKernelGGL(k_init_rep,
                       dim3(grid_for((uint32_t)std::min<size_t>(
                                         total_slots, 0x7FFFFFFF), 256)),
                       dim3(256), 0, s, rep, total_slots, stride,
                       mm ? 1u : 0u);
    return hipGetLastError();
}

hipError_t launch_iota(hipStream_t s, uint32_t* out, uint32_t n) {
    hipLaunchKernelGGL(k_iota, dim3(grid_for(n, 256)), dim3(256), 0, s, out, n);
    return hipGetLastError();
}

hipError_t launch_copy_bytes(hipStream_t s, const uint8_t* blob,
                             const uint64_t* handles, const int64_t* dst_off,
                             uint8_t* out, uint32_t n) {
    CopyBytesParams C{blob, handles, dst_off, out, n};
    hipLaunchKernelGGL(k_copy_bytes, dim3(grid_for(n, 256)), dim3(256), 0, s,
                       C);
    return hipGetLastError();
}

hipError_t launch_ba_offsets(hipStream_t s, const uint8_t* blob,
                             const BaPageDesc* pages, uint32_t n_pages,
                             uint64_t* out, unsigned long long* err_flag) {
    hipLaunchKernelGGL(k_ba_offsets, dim3(grid_for(n_pages, 256)), dim3(256),
                       0, s, blob, pages, n_pages, out, err_flag);
    return hipGetLastError();
}

hipError_t launch_tag_filter(hipStream_t s, const TagFilterParams& f) {
    uint32_t work = (uint32_t)std::min<int64_t>(f.n_rows, 0x7FFFFFFF);
    hipLaunchKernelGGL(k_tag_filter, dim3(grid_for(work, 256)), dim3(256), 0,
                       s, f);
    return hipGetLastError();
}

hipError_t launch_tsid_intersect(hipStream_t s, const uint64_t* a,
                                 unsigned long long n_a, const uint64_t* b,
                                 unsigned long long n_b, uint64_t* out,
                                 unsigned long long* cursor) {
    uint32_t work = (uint32_t)std::min<unsigned long long>(n_a, 0x7FFFFFFF);
    hipLaunchKernelGGL(k_tsid_intersect, dim3(grid_for(work, 256)), dim3(256),
                       0, s, a, n_a, b, n_b, out, cursor);
    return hipGetLastError();
}

hipError_t launch_unique_u64(hipStream_t s, const uint64_t* in,
                             unsigned long long n, uint64_t* out,
                             unsigned long long* cursor) {
    uint32_t work = (uint32_t)std::min<unsigned long long>(n, 0x7FFFFFFF);
    hipLaunchKernelGGL(k_unique_u64, dim3(grid_for(work, 256)), dim3(256), 0,
                       s, in, n, out, cursor);
    return hipGetLastError();
}

hipError_t sort_keys_u64(hipStream_t s, const uint64_t* keys_in,
                         uint64_t* keys_out, size_t n, void** d_temp,
                         size_t* temp_bytes) {
    size_t need = 0;
    hipError_t e = rocprim::radix_sort_keys(nullptr, need, keys_in, keys_out,
                                            n, 0, 64, s);
    if (e != hipSuccess) return e;
    if (need > *temp_bytes) {
        if (*d_temp) (void)hipFree(*d_temp);
        e = hipMalloc(d_temp, need);
        if (e != hipSuccess) { *temp_bytes = 0; *d_temp = nullptr; return e; }
        *temp_bytes = need;
    }
    return rocprim::radix_sort_keys(*d_temp, *temp_bytes, keys_in, keys_out,
                                    n, 0, 64, s);
}

hipError_t sort_pairs_u64(hipStream_t s, const uint64_t* keys_in,
                          uint64_t* keys_out, const uint32_t* vals_in,
                          uint32_t* vals_out, size_t n, void** d_temp,
                          size_t* temp_bytes) {
    size_t need = 0;
    hipError_t e = rocprim::radix_sort_pairs(nullptr, need, keys_in, keys_out,
                                             vals_in, vals_out, n, 0, 64, s);
    if (e != hipSuccess) return e;
    if (need > *temp_bytes) {
        if (*d_temp) hipFree(*d_temp);
        e = hipMalloc(d_temp, need);
        if (e != hipSuccess) { *temp_bytes = 0; *d_temp = nullptr; return e; }
        *temp_bytes = need;
    }
    return rocprim::radix_sort_pairs(*d_temp, *temp_bytes, keys_in, keys_out,
                                     vals_in, vals_out, n, 0, 64, s);
}

}  // namespace hx
