// Per-GPU pool shard for infinistore-amd.
//
// The reference drives one pinned-host-DRAM pool from a single libuv thread
// (/root/reference/src/infinistore.cpp:1, mempool.cpp:29-44). This build
// shards the pool across MI355X GPUs: each shard owns an HBM3E arena
// (hipMalloc), a bitmap allocator, a small pool of HIP streams with
// descriptor-staging slots, and a completion thread that waits on HIP events
// FIFO and hands results back to the event loop (SURVEY.md §7 hard part 4).
// In CPU mode (no GPU) a shard is a host-DRAM arena and copies run inline.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#include "../core/mempool.h"
#include "../gpu/gpu.h"

namespace ifs {

struct ShardOptions {
    int device = -1;          // GPU ordinal; -1 = CPU shard
    size_t pool_bytes = 0;    // initial arena size
    size_t block_granule = 64 << 10;  // bitmap granule (minimal_allocate_size)
    int n_streams = 4;
    // Slots bound the number of in-flight copy jobs; a handler out of slots
    // BLOCKS in acquire_slot, so the count must exceed the worst concurrent
    // request fan-in (64 saturation clients x ~2 in flight), not match it.
    // Bigger requests chunk across slots (submit_copy's chunk loop).
    int slots_per_stream = 48;
    size_t max_descs_per_slot = 16384;  // 16K blocks/chunk -> 128 KiB ptrs/side
    bool auto_extend = false;
    size_t extend_bytes = 10ull << 30;
};

class Shard {
   public:
    // Descriptor list: uniform-size block copies src[i] -> dst[i].
    struct CopyJob {
        // kCopy moves bytes; the fp8 transforms convert while moving
        // (bytes_per_block is the LOGICAL bf16 size in all modes — the fp8
        // side of a transform is half that).
        enum class Xform { kCopy = 0, kQuantBf16Fp8, kDequantFp8Bf16 };
        std::vector<uint64_t> src;
        std::vector<uint64_t> dst;
        size_t bytes_per_block = 0;
        Xform xform = Xform::kCopy;
        // kQuant: per-block scales, written at [scales_off..) before done().
        std::shared_ptr<std::vector<float>> scales_out;
        size_t scales_off = 0;
        // kDequant: per-block scale inputs.
        std::vector<float> scales_in;
        // Completion callback; invoked exactly once from the shard completion
        // thread (GPU) or inline (CPU shard). ok=false means the copy failed.
        std::function<void(bool ok)> done;
    };

    explicit Shard(const ShardOptions& opt);
    ~Shard();
    Shard(const Shard&) = delete;
    Shard& operator=(const Shard&) = delete;

    bool init();  // allocate arena, streams, start completion thread

    int device() const { return opt_.device; }
    bool on_gpu() const { return opt_.device >= 0; }

    // Allocator (thread-safe).
    bool allocate(size_t size, size_t n, const AllocationCallback& cb);
    bool deallocate(void* ptr, size_t size, int pool_idx);
    // Bulk free under ONE allocator-lock hold: a 10k-key delete sweep doing
    // one lock acquisition per block convoyed against concurrent batch
    // allocations (measured 1.3 s per delete at 64 clients).
    struct BlockFree {
        void* ptr;
        size_t size;
        int pool_idx;
    };
    void deallocate_bulk(const std::vector<BlockFree>& frees);
    size_t used_blocks();
    size_t total_blocks();
    bool contains(const void* p);

    // Submit a batched copy; returns false if the job could not be launched
    // (done() is NOT called in that case).
    bool submit_copy(CopyJob&& job);

    // Fabric IO (TCP data path): bulk host<->pool copies run on a dedicated
    // worker thread so IO loops never block on hipMemcpy/memcpy.
    struct FabricJob {
        bool is_put = false;
        std::shared_ptr<std::vector<uint8_t>> host;  // source (put) / dest (get)
        std::shared_ptr<uint8_t[]> raw_host;         // alternative uninit dest
        std::vector<uint64_t> block_ptrs;            // pool block addresses
        std::vector<size_t> host_offsets;            // absolute offsets into *host
        size_t bytes_per_block = 0;
        // Invoked on the worker thread; captured state (body buffers, block
        // refs) is released when the job is destroyed after the call.
        std::function<void(bool)> done;
    };
    void submit_fabric(FabricJob&& job);

    // Extend pool by one arena (called off the hot path).
    // Adds one arena of extend_bytes; reports the new arena's base so the
    // caller can MR-register it on the verbs fabric (the reference registers
    // an MR per pool as pools are created, mempool.cpp:29-44).
    bool extend(void** arena_out = nullptr);
    bool need_extend();

    // Compaction planning: for each (ptr, size), try to find a lower slot in
    // the same pool; marks the new slots used. Returns the planned moves.
    struct Move {
        void* old_ptr;
        void* new_ptr;
        size_t size;
        int pool_idx;
    };
    std::vector<Move> plan_compaction(const std::vector<std::pair<void*, size_t>>& movable);

    // Fragmentation stats (largest contiguous free run in bytes, per pool sum).
    size_t largest_free_run_bytes();

    // Enumerate pool arenas (for fabric MR registration).
    void for_each_arena(const std::function<void(void*, size_t, bool on_gpu)>& fn);

   private:
    // Descriptors live in pinned host memory for every job size: the copy
    // kernels read each block's descriptor once per workgroup (LDS
    // broadcast), so there is no per-unit PCIe tax and no per-job SDMA/blit
    // upload (both variants measured and rejected —
    // profiles/rocprof_bench_r02.txt).
    struct Slot {
        uint64_t* h_src = nullptr;  // pinned
        uint64_t* h_dst = nullptr;
        float* h_scale = nullptr;  // fp8 scales (kernel writes/reads in place)
        gpu::Event event = nullptr;
        bool busy = false;
    };
    struct PendingTask {
        Slot* slot;
        std::function<void(bool)> done;  // may be empty for chunked sub-jobs
        // quant jobs: copy slot->h_scale[0..n) into (*scales_out)[off..)
        // BEFORE the slot is released (the next submitter reuses h_scale).
        std::shared_ptr<std::vector<float>> scales_out;
        size_t scales_off = 0;
        size_t scales_n = 0;
    };
    // Fully per-stream state: slots, pending FIFO, and their synchronization
    // live with the stream, so submitters and the completion thread of one
    // stream never contend with other streams'. (Round 1 used one shard-wide
    // mutex + notify_all; at 64 saturation clients every completion woke ~60
    // blocked handlers and the completion threads' event spin-poll hammered
    // the same lock — task_mu_ was the shard's true bottleneck.)
    struct StreamCtx {
        gpu::Stream stream = nullptr;
        std::mutex mu;
        std::condition_variable slot_cv;  // a slot was freed
        std::condition_variable task_cv;  // a task was enqueued
        std::vector<Slot> slots;
        std::deque<PendingTask> pending;  // FIFO per stream (mu)
        int next_slot = 0;
    };

    Slot* acquire_slot(StreamCtx& sc);
    void completion_loop(size_t stream_idx);
    void fabric_loop();

    ShardOptions opt_;
    MM mm_;
    std::mutex alloc_mu_;

    std::deque<StreamCtx> streams_;  // deque: StreamCtx is not movable (mutex)
    std::atomic<uint32_t> next_stream_{0};

    std::vector<std::thread> completion_threads_;  // one per stream
    std::atomic<bool> stopping_{false};
    bool inited_ = false;

    std::deque<FabricJob> fabric_q_;
    std::mutex fabric_mu_;
    std::condition_variable fabric_cv_;
    std::thread fabric_thread_;

    // Fabric staging (GPU shards): the TCP data path moves payloads through
    // a pinned bounce buffer with ONE hipMemcpyAsync + one scatter/gather
    // kernel launch per ~32 MB chunk, double-buffered so the host-side
    // memcpy of chunk k+1 overlaps the device work of chunk k. Replaces the
    // per-block synchronous hipMemcpy loop (the reference's hot-loop shape,
    // infinistore.cpp:622-625/747-748, which capped the TCP fabric at
    // ~1.5 GB/s).
    static constexpr size_t kFabStageBytes = 32ull << 20;
    static constexpr size_t kFabStageDescs = 8192;
    struct FabBuf {
        uint8_t* h_stage = nullptr;  // pinned payload bounce
        uint8_t* d_stage = nullptr;  // device-side chunk
        uint64_t* h_desc = nullptr;  // pinned [src[descs] | dst[descs]]
        uint64_t* d_desc = nullptr;
        gpu::Event event = nullptr;  // chunk-complete marker
        bool in_flight = false;
        // GET bookkeeping for the deferred host copy-out of this chunk
        size_t out_first = 0, out_n = 0;
    };
    FabBuf fab_[2];
    gpu::Stream fab_stream_ = nullptr;
    bool fab_ready_ = false;
    bool fabric_stage_init();          // fabric thread only
    void fabric_stage_teardown();      // fabric thread only
    bool fabric_run_gpu(FabricJob& job, uint8_t* base);
};

}  // namespace ifs
