// HIP/ROCm layer for infinistore-amd (MI355X-native).
//
// Replaces the reference's CUDA-runtime call sites with an
// MI355X-first design: the pool lives in hipMalloc'd HBM3E, per-block
// hipMemcpyAsync loops are replaced by ONE batched gather/scatter HIP kernel
// launch per request (csrc/gpu/kernels.hip), and completion runs on pooled
// HIP streams with event FIFOs per shard.
//
// Everything here degrades gracefully when no GPU is present (CPU-only
// server mode, BASELINE config 1): `available()` is false and callers fall
// back to host arenas + memcpy.
#pragma once

#include <cstddef>
#include <cstdint>
#include <string>

namespace ifs {
namespace gpu {

bool available();
int device_count();
std::string device_name(int dev);
size_t device_total_mem(int dev);

// --- memory ----------------------------------------------------------------
void* alloc_device(int dev, size_t bytes);           // hipMalloc on device
void free_device(void* p);
void* alloc_host_pinned(size_t bytes);               // hipHostMalloc (falls back to aligned_alloc)
void free_host_pinned(void* p);
bool memcpy_d2h(void* dst, const void* src, size_t n);
bool memcpy_h2d(void* dst, const void* src, size_t n);
bool memcpy_d2d(void* dst, const void* src, size_t n);  // same or peer device
// Async variants on a stream (hipMemcpyAsync).
using Stream = void*;
bool memcpy_h2d_async(void* dst, const void* src, size_t n, Stream s);
bool memcpy_d2h_async(void* dst, const void* src, size_t n, Stream s);
bool memcpy_any_async(void* dst, const void* src, size_t n, Stream s);

// --- IPC -------------------------------------------------------------------
constexpr size_t kIpcHandleSize = 64;  // == HIP_IPC_HANDLE_SIZE
struct IpcHandle {
    uint8_t bytes[kIpcHandleSize];
};

// Export `ptr`'s allocation; also reports the offset of ptr inside that
// allocation so importers can reconstruct the exact address (works with the
// PyTorch caching allocator, unlike the reference which needs
// PYTORCH_NO_CUDA_MEMORY_CACHING).
// alloc_size (optional): the containing allocation's byte size — callers
// must refuse the IPC path for allocations >= 2 GiB (hipIpcOpenMemHandle
// hangs forever importing them under dmabuf IPC on this ROCm/driver stack;
// scripts/ipc_size_probe.py bisected the threshold).
bool ipc_export(const void* ptr, IpcHandle* handle, uint64_t* base_offset,
                uint64_t* alloc_size = nullptr);
void* ipc_open(const IpcHandle& handle, int src_device);
bool ipc_close(void* base);

// Enable xGMI peer access dev -> peer (idempotent). Returns false if the
// pair cannot peer.
bool enable_peer_access(int dev, int peer);

// True if ptr is device (HBM) memory — hipPointerGetAttributes probe, the
// HIP analog of the reference's host-vs-device MR classification.
bool is_device_pointer(const void* ptr);

// Export a dmabuf fd covering [ptr, ptr+size) of device memory (the amdgpu
// path for registering HBM with an RDMA NIC: ibv_reg_dmabuf_mr). Returns
// the fd or -1; *offset is the start offset inside the dmabuf.
int export_dmabuf(void* ptr, size_t size, uint64_t* offset);

// --- streams / events (opaque wrappers so non-HIP TUs can hold them) -------
using Event = void*;
Stream stream_create(int dev);
void stream_destroy(Stream s);
bool stream_sync(Stream s);
Event event_create(int dev);
void event_destroy(Event e);
bool event_record(Event e, Stream s);
bool event_sync(Event e);      // blocking wait
bool event_query(Event e);     // true if complete
bool set_device(int dev);

// --- batched block copy ----------------------------------------------------
// One launch copies n_blocks of `bytes_per_block` from src_ptrs[i] to
// dst_ptrs[i]. The pointer arrays must be device-VISIBLE: device memory or
// pinned host (hipHostMalloc — ROCm unified addressing; each workgroup reads
// its 16 B descriptor once, so keeping descriptors in pinned host memory
// avoids any per-launch SDMA upload). Chooses the vectorized 16 B/lane
// kernel when every pointer and the size are 16-byte aligned, else a
// byte-granular fallback.
bool launch_copy_blocks(int dev, Stream s, const uint64_t* dev_src_ptrs,
                        const uint64_t* dev_dst_ptrs, int n_blocks, size_t bytes_per_block,
                        bool aligned16);

// Small batches (n <= 16, 16B-aligned): descriptors passed as kernel args
// from HOST arrays — no device staging needed.
bool launch_copy_blocks_inline(int dev, Stream s, const uint64_t* src_ptrs,
                               const uint64_t* dst_ptrs, int n_blocks, size_t bytes_per_block);

// --- fp8 KV compression ------------------------------------------------------
// Quantize n blocks of bf16 (elems_per_block each) into fp8 e4m3 blocks of
// half the byte size; one scale (absmax/448) per block -> dev_scales[i].
bool launch_quant_blocks(int dev, Stream s, const uint64_t* dev_src_ptrs,
                         const uint64_t* dev_dst_ptrs, float* dev_scales, int n_blocks,
                         size_t elems_per_block);
// Inverse: fp8 blocks + scales -> bf16 blocks.
bool launch_dequant_blocks(int dev, Stream s, const uint64_t* dev_src_ptrs,
                           const uint64_t* dev_dst_ptrs, const float* dev_scales, int n_blocks,
                           size_t elems_per_block);

// --- block fingerprint ------------------------------------------------------
// 64-bit position-salted fingerprint per block -> out_hashes[i] (device mem).
bool launch_hash_blocks(int dev, Stream s, const uint64_t* dev_ptrs, int n_blocks,
                        size_t bytes_per_block, uint64_t* dev_out_hashes);

const char* last_error();

}  // namespace gpu
}  // namespace ifs
