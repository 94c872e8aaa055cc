// Microbenchmark: LDS atomic vs plain-DS throughput on gfx950 — is the
// aggregate kernel's fixed-rate wall the LDS ATOMIC service rate?
// Each wave hammers a 1024-slot LDS table with hash-scattered ops.
#include <hip/hip_runtime.h>
#include <cstdio>

#define NE 1024

__device__ __forceinline__ uint32_t mix(uint32_t x) {
    x ^= x >> 16; x *= 0x7feb352d; x ^= x >> 15; x *= 0x846ca68b; x ^= x >> 16;
    return x;
}

template <int MODE>
__global__ void __launch_bounds__(256)
k_lds(uint64_t* sink, uint32_t iters) {
    __shared__ double lsum[NE];
    __shared__ unsigned int lcnt[NE];
    for (uint32_t i = threadIdx.x; i < NE; i += blockDim.x) {
        lsum[i] = 0; lcnt[i] = 0;
    }
    __syncthreads();
    uint32_t h = threadIdx.x * 2654435761u + blockIdx.x;
    double acc = 1.0;
    for (uint32_t k = 0; k < iters; k++) {
        h = mix(h + k);
        uint32_t slot = h & (NE - 1);
        if (MODE == 0) {                 // f64 atomic add
            atomicAdd(&lsum[slot], acc);
        } else if (MODE == 1) {          // u32 atomic add
            atomicAdd(&lcnt[slot], 1u);
        } else if (MODE == 2) {          // f64 atomic + u32 atomic (agg pair)
            atomicAdd(&lsum[slot], acc);
            atomicAdd(&lcnt[slot], 1u);
        } else if (MODE == 3) {          // plain read+add+write (racy, bench only)
            double v = lsum[slot];
            lsum[slot] = v + acc;
        } else if (MODE == 4) {          // ds_read only
            acc += lsum[slot];
        } else if (MODE == 5) {          // shfl loop (bpermute cost)
            double v = acc;
            for (int d = 1; d < 8; d <<= 1) v += __shfl_down(v, d, 64);
            acc = v * 1e-30 + 1.0;
        }
    }
    __syncthreads();
    if (threadIdx.x == 0) sink[blockIdx.x] = (uint64_t)(acc + lsum[0] + lcnt[0]);
}

int main() {
    uint64_t* sink;
    (void)hipMalloc(&sink, 4096 * 8);
    const uint32_t iters = 20000;
    const int grid = 2048;   // >> 256 CUs
    const char* names[6] = {"ds_add_f64", "ds_add_u32", "f64+u32 pair",
                            "plain r+w f64", "ds_read f64", "shfl x3"};
    for (int mode = 0; mode < 6; mode++) {
        hipEvent_t a, b;
        (void)hipEventCreate(&a); (void)hipEventCreate(&b);
        auto launch = [&](int m) {
            switch (m) {
            case 0: hipLaunchKernelGGL(k_lds<0>, grid, 256, 0, 0, sink, iters); break;
            case 1: hipLaunchKernelGGL(k_lds<1>, grid, 256, 0, 0, sink, iters); break;
            case 2: hipLaunchKernelGGL(k_lds<2>, grid, 256, 0, 0, sink, iters); break;
            case 3: hipLaunchKernelGGL(k_lds<3>, grid, 256, 0, 0, sink, iters); break;
            case 4: hipLaunchKernelGGL(k_lds<4>, grid, 256, 0, 0, sink, iters); break;
            case 5: hipLaunchKernelGGL(k_lds<5>, grid, 256, 0, 0, sink, iters); break;
            }
        };
        launch(mode);  // warmup
        (void)hipDeviceSynchronize();
        (void)hipEventRecord(a);
        launch(mode);
        (void)hipEventRecord(b);
        (void)hipDeviceSynchronize();
        float ms = 0;
        (void)hipEventElapsedTime(&ms, a, b);
        double ops = double(grid) * 256 * iters;  // per-lane ops
        printf("%-14s %8.3f ms  %8.2f G lane-ops/s  (%5.2f per CU per cycle @2.4GHz)\n",
               names[mode], ms, ops / ms / 1e6,
               ops / (ms * 1e-3) / 256.0 / 2.4e9);
    }
    return 0;
}
