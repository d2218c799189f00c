// Standalone sweep of batched block-copy kernel variants on gfx950:
// finds the best (grid, threads, unroll) for the 2048 x 128 KB request shape.
//   hipcc --offload-arch=gfx950 -O3 scripts/copybench.hip -o /tmp/copybench
//   /tmp/copybench
#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

#define CHECK(x)                                                    \
    if ((x) != hipSuccess) {                                        \
        printf("HIP error %s at %d\n", hipGetErrorString(x), __LINE__); \
        return 1;                                                   \
    }

__global__ void copy_v1(const uint64_t* __restrict__ src_ptrs,
                        const uint64_t* __restrict__ dst_ptrs, int n_blocks,
                        uint64_t units_per_block) {
    uint64_t total = static_cast<uint64_t>(n_blocks) * units_per_block;
    uint64_t stride = static_cast<uint64_t>(gridDim.x) * blockDim.x;
    for (uint64_t u = blockIdx.x * static_cast<uint64_t>(blockDim.x) + threadIdx.x; u < total;
         u += stride) {
        uint64_t b = u / units_per_block;
        uint64_t off = u - b * units_per_block;
        const uint4* s = reinterpret_cast<const uint4*>(src_ptrs[b]) + off;
        uint4* d = reinterpret_cast<uint4*>(dst_ptrs[b]) + off;
        *d = *s;
    }
}

// One workgroup per (block, chunk): no division in the loop; each wave walks
// a contiguous span of one KV block.
template <int UNROLL>
__global__ void copy_v2(const uint64_t* __restrict__ src_ptrs,
                        const uint64_t* __restrict__ dst_ptrs, uint64_t units_per_block,
                        int chunks_per_block) {
    int blk = blockIdx.x / chunks_per_block;
    int chunk = blockIdx.x % chunks_per_block;
    uint64_t per_chunk = units_per_block / chunks_per_block;
    const uint4* s = reinterpret_cast<const uint4*>(src_ptrs[blk]) + chunk * per_chunk;
    uint4* d = reinterpret_cast<uint4*>(dst_ptrs[blk]) + chunk * per_chunk;
    for (uint64_t u = threadIdx.x * UNROLL; u + UNROLL <= per_chunk;
         u += blockDim.x * UNROLL) {
#pragma unroll
        for (int k = 0; k < UNROLL; k++) d[u + k] = s[u + k];
    }
}

// v3: like v1 but 2 uint4 per lane per iteration (32 B), shift instead of div
// (units_per_block is a power of two for 128 KB blocks).
__global__ void copy_v3(const uint64_t* __restrict__ src_ptrs,
                        const uint64_t* __restrict__ dst_ptrs, int n_blocks, int log2_upb) {
    uint64_t upb = 1ull << log2_upb;
    uint64_t total = (static_cast<uint64_t>(n_blocks) << log2_upb) / 2;
    uint64_t stride = static_cast<uint64_t>(gridDim.x) * blockDim.x;
    for (uint64_t p = blockIdx.x * static_cast<uint64_t>(blockDim.x) + threadIdx.x; p < total;
         p += stride) {
        uint64_t u = p * 2;
        uint64_t b = u >> log2_upb;
        uint64_t off = u & (upb - 1);
        const uint4* s = reinterpret_cast<const uint4*>(src_ptrs[b]) + off;
        uint4* d = reinterpret_cast<uint4*>(dst_ptrs[b]) + off;
        uint4 a0 = s[0], a1 = s[1];
        d[0] = a0;
        d[1] = a1;
    }
}

int main() {
    const int NB = 2048;
    const size_t BS = 128 << 10;
    const uint64_t UPB = BS / 16;
    uint8_t *pool, *client;
    CHECK(hipMalloc(&pool, NB * BS));
    CHECK(hipMalloc(&client, NB * BS));
    CHECK(hipMemset(pool, 1, NB * BS));
    std::vector<uint64_t> hs(NB), hd(NB);
    for (int i = 0; i < NB; i++) {
        hs[i] = reinterpret_cast<uint64_t>(client + i * BS);
        hd[i] = reinterpret_cast<uint64_t>(pool + i * BS);
    }
    uint64_t *ds, *dd;
    CHECK(hipMalloc(&ds, NB * 8));
    CHECK(hipMalloc(&dd, NB * 8));
    CHECK(hipMemcpy(ds, hs.data(), NB * 8, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(dd, hd.data(), NB * 8, hipMemcpyHostToDevice));

    hipEvent_t e0, e1;
    CHECK(hipEventCreate(&e0));
    CHECK(hipEventCreate(&e1));
    auto bench = [&](const char* name, auto launch) {
        launch();  // warmup
        CHECK(hipDeviceSynchronize());
        CHECK(hipEventRecord(e0));
        for (int r = 0; r < 10; r++) launch();
        CHECK(hipEventRecord(e1));
        CHECK(hipEventSynchronize(e1));
        float ms;
        CHECK(hipEventElapsedTime(&ms, e0, e1));
        double gbps = 10.0 * NB * BS / (ms / 1e3) / 1e9;
        printf("%-28s %8.1f us  payload %7.0f GB/s (HBM r+w %7.0f GB/s)\n", name,
               ms * 100, gbps, 2 * gbps);
        return 0;
    };

    for (int grid : {2048, 4096, 8192}) {
        for (int thr : {256, 512}) {
            char nm[64];
            snprintf(nm, sizeof(nm), "v1 grid=%d thr=%d", grid, thr);
            bench(nm, [&] {
                hipLaunchKernelGGL(copy_v1, dim3(grid), dim3(thr), 0, 0, ds, dd, NB, UPB);
            });
        }
    }
    for (int cpb : {1, 2, 4, 8}) {
        for (int thr : {256, 512, 1024}) {
            char nm[64];
            snprintf(nm, sizeof(nm), "v2 u4 cpb=%d thr=%d", cpb, thr);
            bench(nm, [&] {
                hipLaunchKernelGGL(copy_v2<4>, dim3(NB * cpb), dim3(thr), 0, 0, ds, dd, UPB,
                                   cpb);
            });
        }
    }
    for (int grid : {2048, 4096, 8192}) {
        char nm[64];
        snprintf(nm, sizeof(nm), "v3 32B grid=%d thr=256", grid);
        bench(nm, [&] {
            hipLaunchKernelGGL(copy_v3, dim3(grid), dim3(256), 0, 0, ds, dd, NB, 13);
        });
    }
    // plain hipMemcpyDtoD for reference
    bench("hipMemcpy D2D (whole range)", [&] {
        hipMemcpyAsync(pool, client, NB * BS, hipMemcpyDeviceToDevice, 0);
    });
    return 0;
}
