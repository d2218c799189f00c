// MI355X (gfx950) HIP implementation: arenas, IPC, streams, and the batched
// block gather/scatter + fingerprint kernels. See gpu.h for the design notes.
//
// Kernels are written for CDNA4: 64-wide wavefronts, 16 B/lane vectorized
// global accesses (uint4), grid sized to cover 256 CUs with grid-stride
// loops (memory-bound kernel rule from the CDNA4 programming guide).
#include "gpu.h"

#include <hip/hip_runtime.h>

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>

#include "../core/log.h"

namespace ifs {
namespace gpu {

static thread_local char g_err[256] = {0};
const char* last_error() { return g_err; }

static bool fail(const char* what, hipError_t e) {
    snprintf(g_err, sizeof(g_err), "%s: %s", what, hipGetErrorString(e));
    ERROR("%s", g_err);
    return false;
}

#define HIP_OK(call)                                  \
    do {                                              \
        hipError_t _e = (call);                       \
        if (_e != hipSuccess) return fail(#call, _e); \
    } while (0)

static int cached_device_count() {
    static int n = []() {
        int c = 0;
        hipError_t e = hipGetDeviceCount(&c);
        if (e != hipSuccess) return 0;
        return c;
    }();
    return n;
}

bool available() { return cached_device_count() > 0; }
int device_count() { return cached_device_count(); }

std::string device_name(int dev) {
    hipDeviceProp_t prop;
    if (hipGetDeviceProperties(&prop, dev) != hipSuccess) return "unknown";
    return prop.name;
}

size_t device_total_mem(int dev) {
    hipDeviceProp_t prop;
    if (hipGetDeviceProperties(&prop, dev) != hipSuccess) return 0;
    return prop.totalGlobalMem;
}

bool set_device(int dev) {
    HIP_OK(hipSetDevice(dev));
    return true;
}

void* alloc_device(int dev, size_t bytes) {
    if (hipSetDevice(dev) != hipSuccess) return nullptr;
    void* p = nullptr;
    hipError_t e = hipMalloc(&p, bytes);
    if (e != hipSuccess) {
        fail("hipMalloc", e);
        return nullptr;
    }
    return p;
}

void free_device(void* p) {
    if (p) hipFree(p);
}

void* alloc_host_pinned(size_t bytes) {
    if (available()) {
        void* p = nullptr;
        if (hipHostMalloc(&p, bytes, hipHostMallocDefault) == hipSuccess) return p;
    }
    void* p = nullptr;
    if (posix_memalign(&p, 4096, bytes) != 0) return nullptr;
    return p;
}

void free_host_pinned(void* p) {
    if (!p) return;
    if (available()) {
        if (hipHostFree(p) == hipSuccess) return;
    }
    free(p);
}

bool memcpy_d2h(void* dst, const void* src, size_t n) {
    HIP_OK(hipMemcpy(dst, src, n, hipMemcpyDeviceToHost));
    return true;
}
bool memcpy_h2d(void* dst, const void* src, size_t n) {
    HIP_OK(hipMemcpy(dst, src, n, hipMemcpyHostToDevice));
    return true;
}
bool memcpy_d2d(void* dst, const void* src, size_t n) {
    HIP_OK(hipMemcpy(dst, src, n, hipMemcpyDefault));
    return true;
}
bool memcpy_h2d_async(void* dst, const void* src, size_t n, Stream s) {
    HIP_OK(hipMemcpyAsync(dst, src, n, hipMemcpyHostToDevice, reinterpret_cast<hipStream_t>(s)));
    return true;
}
bool memcpy_d2h_async(void* dst, const void* src, size_t n, Stream s) {
    HIP_OK(hipMemcpyAsync(dst, src, n, hipMemcpyDeviceToHost, reinterpret_cast<hipStream_t>(s)));
    return true;
}
bool memcpy_any_async(void* dst, const void* src, size_t n, Stream s) {
    HIP_OK(hipMemcpyAsync(dst, src, n, hipMemcpyDefault, reinterpret_cast<hipStream_t>(s)));
    return true;
}

// --- IPC -------------------------------------------------------------------
static_assert(sizeof(hipIpcMemHandle_t) == kIpcHandleSize, "hip ipc handle size");

bool ipc_export(const void* ptr, IpcHandle* handle, uint64_t* base_offset,
                uint64_t* alloc_size) {
    // Find the allocation base so tensors offset inside a caching-allocator
    // segment still round-trip exactly.
    void* base = nullptr;
    hipError_t e = hipPointerGetAttribute(&base, HIP_POINTER_ATTRIBUTE_RANGE_START_ADDR,
                                          reinterpret_cast<hipDeviceptr_t>(const_cast<void*>(ptr)));
    if (e != hipSuccess || base == nullptr) {
        base = const_cast<void*>(ptr);  // assume ptr is the base
    }
    *base_offset = reinterpret_cast<uintptr_t>(ptr) - reinterpret_cast<uintptr_t>(base);
    if (alloc_size) {
        size_t sz = 0;
        if (hipPointerGetAttribute(&sz, HIP_POINTER_ATTRIBUTE_RANGE_SIZE,
                                   reinterpret_cast<hipDeviceptr_t>(base)) != hipSuccess)
            sz = 0;  // unknown: caller decides
        *alloc_size = sz;
    }
    hipIpcMemHandle_t h;
    HIP_OK(hipIpcGetMemHandle(&h, base));
    memcpy(handle->bytes, &h, sizeof(h));
    return true;
}

void* ipc_open(const IpcHandle& handle, int src_device) {
    if (hipSetDevice(src_device) != hipSuccess) return nullptr;
    hipIpcMemHandle_t h;
    memcpy(&h, handle.bytes, sizeof(h));
    void* p = nullptr;
    hipError_t e = hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess);
    if (e != hipSuccess) {
        fail("hipIpcOpenMemHandle", e);
        return nullptr;
    }
    return p;
}

bool ipc_close(void* base) {
    HIP_OK(hipIpcCloseMemHandle(base));
    return true;
}

bool is_device_pointer(const void* ptr) {
    hipPointerAttribute_t attr;
    if (hipPointerGetAttributes(&attr, ptr) != hipSuccess) {
        (void)hipGetLastError();  // clear sticky error for untracked host ptrs
        return false;
    }
    return attr.type == hipMemoryTypeDevice;
}

int export_dmabuf(void* ptr, size_t size, uint64_t* offset) {
    *offset = 0;
#if HIP_VERSION >= 60000000 || defined(hipMemRangeHandleTypeDmaBufFd)
    int fd = -1;
    hipError_t e = hipMemGetHandleForAddressRange(&fd, reinterpret_cast<hipDeviceptr_t>(ptr),
                                                  size, hipMemRangeHandleTypeDmaBufFd, 0);
    if (e != hipSuccess) {
        fail("hipMemGetHandleForAddressRange", e);
        return -1;
    }
    return fd;
#else
    (void)ptr;
    (void)size;
    return -1;
#endif
}

bool enable_peer_access(int dev, int peer) {
    if (dev == peer) return true;
    int can = 0;
    HIP_OK(hipDeviceCanAccessPeer(&can, dev, peer));
    if (!can) return false;
    HIP_OK(hipSetDevice(dev));
    hipError_t e = hipDeviceEnablePeerAccess(peer, 0);
    if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) return fail("enable_peer", e);
    return true;
}

// --- streams / events ------------------------------------------------------
Stream stream_create(int dev) {
    if (hipSetDevice(dev) != hipSuccess) return nullptr;
    hipStream_t s = nullptr;
    if (hipStreamCreateWithFlags(&s, hipStreamNonBlocking) != hipSuccess) return nullptr;
    return reinterpret_cast<Stream>(s);
}
void stream_destroy(Stream s) {
    if (s) hipStreamDestroy(reinterpret_cast<hipStream_t>(s));
}
bool stream_sync(Stream s) {
    HIP_OK(hipStreamSynchronize(reinterpret_cast<hipStream_t>(s)));
    return true;
}
Event event_create(int dev) {
    if (hipSetDevice(dev) != hipSuccess) return nullptr;
    hipEvent_t e = nullptr;
    if (hipEventCreateWithFlags(&e, hipEventDisableTiming) != hipSuccess) return nullptr;
    return reinterpret_cast<Event>(e);
}
void event_destroy(Event e) {
    if (e) hipEventDestroy(reinterpret_cast<hipEvent_t>(e));
}
bool event_record(Event e, Stream s) {
    HIP_OK(hipEventRecord(reinterpret_cast<hipEvent_t>(e), reinterpret_cast<hipStream_t>(s)));
    return true;
}
bool event_sync(Event e) {
    HIP_OK(hipEventSynchronize(reinterpret_cast<hipEvent_t>(e)));
    return true;
}
bool event_query(Event e) {
    return hipEventQuery(reinterpret_cast<hipEvent_t>(e)) == hipSuccess;
}

// ---------------------------------------------------------------------------
// Kernels
// ---------------------------------------------------------------------------

// Batched block copy, vectorized: 16 B per lane per iteration. Replaces the
// reference's per-block cudaMemcpyAsync hot loop
// (/root/reference/src/infinistore.cpp:622-625, 747-748) with one launch per
// request. Lanes walk consecutive uint4 units, so a wave issues fully
// coalesced 1 KiB transactions; grid-stride covers all blocks.
// 2D mapping: workgroup (b, chunk) covers block b's units
// [chunk*T, ...] striding by chunks*T. Each workgroup reads its block's
// descriptor ONCE (lane 0 -> LDS broadcast), so the descriptor arrays can
// live in pinned HOST memory with negligible PCIe cost — the earlier
// per-unit desc[b] indexing either taxed PCIe at ~payload/64 (pinned) or
// forced per-job uploads that rocclr ran as single-workgroup blits at
// ~110 µs each (profiles/rocprof_bench_r02.txt).
__global__ void copy_blocks_vec_kernel(const uint64_t* __restrict__ src_ptrs,
                                       const uint64_t* __restrict__ dst_ptrs,
                                       uint32_t chunks_per_block,
                                       uint64_t units_per_block) {
    uint32_t b = blockIdx.x / chunks_per_block;
    uint32_t chunk = blockIdx.x % chunks_per_block;
    __shared__ uint64_t sd[2];
    if (threadIdx.x == 0) {
        sd[0] = src_ptrs[b];
        sd[1] = dst_ptrs[b];
    }
    __syncthreads();
    const uint4* s = reinterpret_cast<const uint4*>(sd[0]);
    uint4* d = reinterpret_cast<uint4*>(sd[1]);
    uint64_t stride = static_cast<uint64_t>(chunks_per_block) * blockDim.x;
    for (uint64_t u = static_cast<uint64_t>(chunk) * blockDim.x + threadIdx.x;
         u < units_per_block; u += stride)
        d[u] = s[u];
}

// Small-request variant: descriptors passed by value as kernel arguments —
// no staging H2D, shaving two async copies off the single-block latency path.
struct InlineDescs {
    uint64_t src[16];
    uint64_t dst[16];
};

__global__ void copy_blocks_vec_inline_kernel(InlineDescs descs, int n_blocks,
                                              uint64_t units_per_block) {
    uint64_t total = static_cast<uint64_t>(n_blocks) * units_per_block;
    uint64_t stride = static_cast<uint64_t>(gridDim.x) * blockDim.x;
    for (uint64_t u = blockIdx.x * static_cast<uint64_t>(blockDim.x) + threadIdx.x; u < total;
         u += stride) {
        uint64_t b = u / units_per_block;
        uint64_t off = u - b * units_per_block;
        const uint4* s = reinterpret_cast<const uint4*>(descs.src[b]) + off;
        uint4* d = reinterpret_cast<uint4*>(descs.dst[b]) + off;
        *d = *s;
    }
}

bool launch_copy_blocks_inline(int dev, Stream stream, const uint64_t* src_ptrs,
                               const uint64_t* dst_ptrs, int n_blocks, size_t bytes_per_block) {
    if (n_blocks <= 0 || n_blocks > 16 || bytes_per_block % 16 != 0) return false;
    HIP_OK(hipSetDevice(dev));
    hipStream_t s = reinterpret_cast<hipStream_t>(stream);
    InlineDescs descs;
    for (int i = 0; i < n_blocks; i++) {
        descs.src[i] = src_ptrs[i];
        descs.dst[i] = dst_ptrs[i];
    }
    uint64_t upb = bytes_per_block / 16;
    uint64_t total = static_cast<uint64_t>(n_blocks) * upb;
    const int threads = 256;
    int grid = static_cast<int>(std::min<uint64_t>((total + threads - 1) / threads, 4096));
    hipLaunchKernelGGL(copy_blocks_vec_inline_kernel, dim3(grid), dim3(threads), 0, s, descs,
                       n_blocks, upb);
    HIP_OK(hipGetLastError());
    return true;
}

// Byte-granular fallback for unaligned pointers / sizes (same 2D mapping).
__global__ void copy_blocks_byte_kernel(const uint64_t* __restrict__ src_ptrs,
                                        const uint64_t* __restrict__ dst_ptrs,
                                        uint32_t chunks_per_block,
                                        uint64_t bytes_per_block) {
    uint32_t b = blockIdx.x / chunks_per_block;
    uint32_t chunk = blockIdx.x % chunks_per_block;
    __shared__ uint64_t sd[2];
    if (threadIdx.x == 0) {
        sd[0] = src_ptrs[b];
        sd[1] = dst_ptrs[b];
    }
    __syncthreads();
    const uint8_t* s = reinterpret_cast<const uint8_t*>(sd[0]);
    uint8_t* d = reinterpret_cast<uint8_t*>(sd[1]);
    uint64_t stride = static_cast<uint64_t>(chunks_per_block) * blockDim.x;
    for (uint64_t u = static_cast<uint64_t>(chunk) * blockDim.x + threadIdx.x;
         u < bytes_per_block; u += stride)
        d[u] = s[u];
}

// Enough workgroups to fill 256 CUs across 8 XCDs with headroom, but ONE
// descriptor read per workgroup.
static uint32_t copy_chunks_per_block(uint64_t units_per_block, int n_blocks, int threads) {
    uint64_t max_chunks = (units_per_block + threads - 1) / static_cast<uint64_t>(threads);
    uint64_t want = (4096 + n_blocks - 1) / static_cast<uint64_t>(n_blocks);
    uint64_t c = std::min<uint64_t>(std::max<uint64_t>(want, 1), max_chunks);
    return static_cast<uint32_t>(std::max<uint64_t>(c, 1));
}

bool launch_copy_blocks(int dev, Stream stream, const uint64_t* dev_src_ptrs,
                        const uint64_t* dev_dst_ptrs, int n_blocks, size_t bytes_per_block,
                        bool aligned16) {
    if (n_blocks <= 0 || bytes_per_block == 0) return true;
    HIP_OK(hipSetDevice(dev));
    hipStream_t s = reinterpret_cast<hipStream_t>(stream);
    // threads=512 measured fastest for the grid-stride uint4 copy on gfx950
    // (90.7 us vs 93.4 us at 256 for 2048x128 KB; scripts/copybench.hip —
    // also faster than hipMemcpy D2D at 98.2 us).
    const int threads = 512;
    if (aligned16 && bytes_per_block % 16 == 0) {
        uint64_t upb = bytes_per_block / 16;
        uint32_t chunks = copy_chunks_per_block(upb, n_blocks, threads);
        dim3 grid(static_cast<uint32_t>(n_blocks) * chunks);
        hipLaunchKernelGGL(copy_blocks_vec_kernel, grid, dim3(threads), 0, s, dev_src_ptrs,
                           dev_dst_ptrs, chunks, upb);
    } else {
        uint32_t chunks = copy_chunks_per_block(bytes_per_block, n_blocks, threads);
        dim3 grid(static_cast<uint32_t>(n_blocks) * chunks);
        hipLaunchKernelGGL(copy_blocks_byte_kernel, grid, dim3(threads), 0, s, dev_src_ptrs,
                           dev_dst_ptrs, chunks, bytes_per_block);
    }
    HIP_OK(hipGetLastError());
    return true;
}

// --- fp8 KV-cache compression ----------------------------------------------
// MI355X-native extension (no reference equivalent): pages enter the pool
// quantized bf16 -> OCP fp8 e4m3 with ONE power-agnostic scale per page
// (absmax/448), halving HBM footprint per cached page; reads dequantize
// back to bf16 on the way out. Both kernels are memory-bound streaming
// passes: one 512-thread workgroup per page, vectorized 8-elems-per-lane.
// Encoding is done in integer bit math (round-to-nearest-even on the
// normal path) so results match torch.float8_e4m3fn casts to 1 ulp.

__device__ __forceinline__ uint8_t f32_to_e4m3(float f) {
    uint32_t bits = __float_as_uint(f);
    uint8_t sign = static_cast<uint8_t>((bits >> 24) & 0x80);
    uint32_t u = bits & 0x7fffffffu;
    if (u >= 0x43e00000u) return sign | 0x7e;  // |x| >= 448 -> clamp to max
    int e = static_cast<int>(u >> 23) - 127;
    if (e < -6) {  // subnormal region: unit 2^-9
        if (e < -10) return sign;  // rounds to zero
        uint32_t full = 0x800000u | (u & 0x7fffffu);  // 1.m (24-bit)
        int shift = 23 - (e + 9);                     // 2^-9 units
        uint32_t half = 1u << (shift - 1);
        uint32_t q = (full + half) >> shift;          // round half away
        if (q > 8) q = 8;                             // 8 == min normal code
        return sign | static_cast<uint8_t>(q);
    }
    // RNE at mantissa bit 20.
    uint32_t rounded = u + 0x0007ffffu + ((u >> 20) & 1u);
    if (rounded >= 0x43e00000u) return sign | 0x7e;
    e = static_cast<int>(rounded >> 23) - 127;
    uint32_t m = (rounded >> 20) & 7u;
    return sign | static_cast<uint8_t>(((e + 7) << 3) | m);
}

__device__ __forceinline__ float e4m3_to_f32(uint8_t v) {
    uint32_t e = (v >> 3) & 0xfu;
    uint32_t m = v & 7u;
    float mag;
    if (e == 0)
        mag = static_cast<float>(m) * 0.001953125f;  // m * 2^-9
    else
        mag = __uint_as_float(((e - 7 + 127) << 23) | (m << 20));
    return (v & 0x80) ? -mag : mag;
}

// One workgroup per page. Pass 1: workgroup absmax (wavefront shuffle
// reduce + LDS). Pass 2: scale + convert, 8 bf16 in (one uint4) -> 8 fp8
// out (one uint2) per lane per iteration.
__global__ void quant_blocks_bf16_fp8_kernel(const uint64_t* __restrict__ src_ptrs,
                                             const uint64_t* __restrict__ dst_ptrs,
                                             float* __restrict__ scales,
                                             uint64_t elems_per_block) {
    const uint16_t* src = reinterpret_cast<const uint16_t*>(src_ptrs[blockIdx.x]);
    uint8_t* dst = reinterpret_cast<uint8_t*>(dst_ptrs[blockIdx.x]);
    __shared__ float red[8];  // 512 threads / 64-wide wavefronts
    float m = 0.f;
    uint64_t n8 = elems_per_block / 8;
    const uint4* src4 = reinterpret_cast<const uint4*>(src);
    for (uint64_t u = threadIdx.x; u < n8; u += blockDim.x) {
        uint4 pk = src4[u];
        const uint16_t* h = reinterpret_cast<const uint16_t*>(&pk);
#pragma unroll
        for (int i = 0; i < 8; i++) {
            // bf16 absmax == f32 absmax of the widened bits.
            float f = __uint_as_float(static_cast<uint32_t>(h[i]) << 16);
            m = fmaxf(m, fabsf(f));
        }
    }
    for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_down(m, off, 64));
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = m;
    __syncthreads();
    if (threadIdx.x == 0) {
        float mm = red[0];
        for (int w = 1; w < static_cast<int>(blockDim.x) / 64; w++) mm = fmaxf(mm, red[w]);
        red[0] = mm > 0.f ? mm / 448.f : 1.f;  // scale
        scales[blockIdx.x] = red[0];
    }
    __syncthreads();
    float inv = 1.f / red[0];
    for (uint64_t u = threadIdx.x; u < n8; u += blockDim.x) {
        uint4 pk = src4[u];
        const uint16_t* h = reinterpret_cast<const uint16_t*>(&pk);
        uint8_t out[8];
#pragma unroll
        for (int i = 0; i < 8; i++) {
            float f = __uint_as_float(static_cast<uint32_t>(h[i]) << 16);
            out[i] = f32_to_e4m3(f * inv);
        }
        reinterpret_cast<uint2*>(dst)[u] = *reinterpret_cast<const uint2*>(out);
    }
}

// Dequantize fp8 pages back into bf16 client memory (round-to-nearest-even
// bf16 truncation): 8 fp8 in (uint2) -> 8 bf16 out (uint4) per lane.
__global__ void dequant_blocks_fp8_bf16_kernel(const uint64_t* __restrict__ src_ptrs,
                                               const uint64_t* __restrict__ dst_ptrs,
                                               const float* __restrict__ scales,
                                               uint64_t elems_per_block) {
    const uint8_t* src = reinterpret_cast<const uint8_t*>(src_ptrs[blockIdx.x]);
    uint16_t* dst = reinterpret_cast<uint16_t*>(dst_ptrs[blockIdx.x]);
    float scale = scales[blockIdx.x];
    uint64_t n8 = elems_per_block / 8;
    for (uint64_t u = threadIdx.x; u < n8; u += blockDim.x) {
        uint2 pk = reinterpret_cast<const uint2*>(src)[u];
        const uint8_t* b = reinterpret_cast<const uint8_t*>(&pk);
        uint16_t out[8];
#pragma unroll
        for (int i = 0; i < 8; i++) {
            float f = e4m3_to_f32(b[i]) * scale;
            uint32_t fb = __float_as_uint(f);
            fb += 0x7fffu + ((fb >> 16) & 1u);  // RNE to bf16
            out[i] = static_cast<uint16_t>(fb >> 16);
        }
        reinterpret_cast<uint4*>(dst)[u] = *reinterpret_cast<const uint4*>(out);
    }
}

bool launch_quant_blocks(int dev, Stream stream, const uint64_t* dev_src_ptrs,
                         const uint64_t* dev_dst_ptrs, float* dev_scales, int n_blocks,
                         size_t elems_per_block) {
    if (n_blocks <= 0) return true;
    if (elems_per_block % 8 != 0) return false;
    HIP_OK(hipSetDevice(dev));
    hipStream_t s = reinterpret_cast<hipStream_t>(stream);
    hipLaunchKernelGGL(quant_blocks_bf16_fp8_kernel, dim3(n_blocks), dim3(512), 0, s,
                       dev_src_ptrs, dev_dst_ptrs, dev_scales, elems_per_block);
    HIP_OK(hipGetLastError());
    return true;
}

bool launch_dequant_blocks(int dev, Stream stream, const uint64_t* dev_src_ptrs,
                           const uint64_t* dev_dst_ptrs, const float* dev_scales, int n_blocks,
                           size_t elems_per_block) {
    if (n_blocks <= 0) return true;
    if (elems_per_block % 8 != 0) return false;
    HIP_OK(hipSetDevice(dev));
    hipStream_t s = reinterpret_cast<hipStream_t>(stream);
    hipLaunchKernelGGL(dequant_blocks_fp8_bf16_kernel, dim3(n_blocks), dim3(512), 0, s,
                       dev_src_ptrs, dev_dst_ptrs, dev_scales, elems_per_block);
    HIP_OK(hipGetLastError());
    return true;
}

// --- fingerprint -----------------------------------------------------------
// Position-salted 64-bit mix (splitmix64 finalizer); XOR-combined across a
// block so the reduction is order-free, deterministic, and parallel.
__device__ __forceinline__ uint64_t mix64(uint64_t x) {
    x += 0x9e3779b97f4a7c15ull;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
    return x ^ (x >> 31);
}

// One workgroup (256 threads) per KV block.
__global__ void hash_blocks_kernel(const uint64_t* __restrict__ ptrs, uint64_t bytes_per_block,
                                   uint64_t* __restrict__ out) {
    __shared__ uint64_t lds[256];
    int blk = blockIdx.x;
    const uint8_t* base = reinterpret_cast<const uint8_t*>(ptrs[blk]);
    uint64_t n_words = bytes_per_block / 8;
    uint64_t h = 0;
    const uint64_t* w = reinterpret_cast<const uint64_t*>(base);
    for (uint64_t i = threadIdx.x; i < n_words; i += blockDim.x) {
        h ^= mix64(w[i] ^ (i * 0xff51afd7ed558ccdull));
    }
    // Tail bytes (thread 0 only).
    if (threadIdx.x == 0) {
        uint64_t tail = 0;
        uint64_t tb = bytes_per_block - n_words * 8;
        for (uint64_t i = 0; i < tb; i++)
            tail |= static_cast<uint64_t>(base[n_words * 8 + i]) << (8 * i);
        if (tb) h ^= mix64(tail ^ (n_words * 0xff51afd7ed558ccdull));
    }
    lds[threadIdx.x] = h;
    __syncthreads();
    for (int s2 = blockDim.x / 2; s2 > 0; s2 >>= 1) {
        if (threadIdx.x < static_cast<unsigned>(s2)) lds[threadIdx.x] ^= lds[threadIdx.x + s2];
        __syncthreads();
    }
    if (threadIdx.x == 0) out[blk] = mix64(lds[0] ^ bytes_per_block);
}

bool launch_hash_blocks(int dev, Stream stream, const uint64_t* dev_ptrs, int n_blocks,
                        size_t bytes_per_block, uint64_t* dev_out_hashes) {
    if (n_blocks <= 0) return true;
    HIP_OK(hipSetDevice(dev));
    hipStream_t s = reinterpret_cast<hipStream_t>(stream);
    hipLaunchKernelGGL(hash_blocks_kernel, dim3(n_blocks), dim3(256), 0, s, dev_ptrs,
                       bytes_per_block, dev_out_hashes);
    HIP_OK(hipGetLastError());
    return true;
}

}  // namespace gpu
}  // namespace ifs
