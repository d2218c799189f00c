// Block memory pool for infinistore-amd.
//
// Same role as the reference's bitmap pool (cf. /root/reference/src/mempool.h:56-84:
// MM multi-pool manager, first-fit bitmap, auto-extend) but redesigned:
//  * The pool is arena-agnostic: the arena may be MI355X HBM3E (hipMalloc,
//    sized toward 288 GB/GPU) or host DRAM (CPU mode / staging). The
//    reference only supported pinned host DRAM.
//  * Two-level bitmap: a summary bitmap marks fully-used words so first-fit
//    skips 4096 blocks per summary word — at 2.25M blocks/GPU
//    (288 GB / 128 KB) a linear scan of one flat bitmap degrades
//    (SURVEY.md §7 hard part 5).
//  * Free does not reset the scan cursor; freed space is found via the
//    summary instead (the reference's reset-on-free heuristic re-scans from
//    the start of the pool after every free).
#pragma once

#include <cstddef>
#include <cstdint>
#include <functional>
#include <memory>
#include <string>
#include <vector>

namespace ifs {

class MemoryPool {
   public:
    // base: arena base pointer (device or host memory — the pool only does
    // pointer arithmetic, never dereferences). block_size: allocation granule.
    MemoryPool(void* base, size_t size, size_t block_size, int pool_idx);

    // Allocate `size` bytes (rounded up to whole blocks, contiguous).
    // Returns nullptr if no run of free blocks is large enough.
    void* allocate(size_t size);
    // Batch fast path: allocate n pages of `size` as ONE contiguous run
    // (one bitmap scan instead of n). Returns the base or nullptr; pages sit
    // at base + i*ceil(size/block)*block and may be freed individually.
    void* allocate_contiguous(size_t size, size_t n);
    // Compaction helper: allocate only from block range [0, limit_block).
    void* allocate_below(size_t size, size_t limit_block);
    // Returns false on invalid pointer / double free.
    bool deallocate(void* ptr, size_t size);

    size_t block_index_of(void* ptr) const {
        return (reinterpret_cast<uintptr_t>(ptr) - reinterpret_cast<uintptr_t>(base_)) /
               block_size_;
    }
    // Largest contiguous free run, in blocks.
    size_t largest_free_run() const;

    bool contains(void* ptr) const {
        auto p = reinterpret_cast<uintptr_t>(ptr);
        auto b = reinterpret_cast<uintptr_t>(base_);
        return p >= b && p < b + size_;
    }

    size_t total_blocks() const { return n_blocks_; }
    size_t used_blocks() const { return used_blocks_; }
    size_t block_size() const { return block_size_; }
    void* base() const { return base_; }
    size_t size() const { return size_; }
    int pool_idx() const { return pool_idx_; }

    // For compaction planning: snapshot of the used-bitmap.
    const std::vector<uint64_t>& bitmap() const { return bits_; }

   private:
    bool run_is_free(size_t start, size_t nb) const;
    void mark(size_t start, size_t nb, bool used);
    bool find_run(size_t nb, size_t* out_start);
    bool find_run_in(size_t nb, size_t w_begin, size_t w_end, size_t block_limit,
                     size_t* out_start);

    void* base_;
    size_t size_;
    size_t block_size_;
    size_t n_blocks_;
    size_t n_words_;
    size_t used_blocks_ = 0;
    size_t cursor_ = 0;  // word index where the next search starts
    int pool_idx_;
    std::vector<uint64_t> bits_;     // 1 = used
    std::vector<uint64_t> summary_;  // bit j of word i: word i*64+j fully used

    // Single-block LIFO free list: the KV-store steady state frees and
    // re-allocates one-granule pages at high rate (64-client churn measured
    // 25 ms/request in first-fit — SURVEY §7 hard part 5); popping a cached
    // index is O(1). Entries are HINTS — the bitmap stays authoritative and
    // a popped index whose bit is set (stolen by a contiguous run) is
    // dropped. deallocate() pushes single-granule frees.
    std::vector<uint32_t> free_stack_;
    // Contiguous-run scans over a fragmented pool cost O(pool) and fail;
    // after a failure, skip further contiguous attempts until enough frees
    // accumulated to plausibly open a run again.
    bool contig_ok_ = true;
    size_t frees_since_contig_fail_ = 0;
};

// Allocation callback: (ptr, pool_idx) per block — mirrors the reference's
// AllocationCallback (mempool.h:17-19) minus lkey/rkey, which are owned by
// the fabric layer here (the pool is transport-agnostic).
using AllocationCallback = std::function<void(void*, int)>;

// Multi-pool manager: one per shard. Pools share a block size; allocation
// cascades across pools; `need_extend` turns true when the last pool crosses
// the usage ratio so the shard can hipMalloc another arena off the hot path.
class MM {
   public:
    static constexpr double kBlockUsageRatio = 0.8;

    // Takes ownership bookkeeping of arenas via the free_fn.
    using ArenaFree = std::function<void(void*, size_t)>;

    MM() = default;
    ~MM();
    MM(const MM&) = delete;
    MM& operator=(const MM&) = delete;

    int add_pool(void* base, size_t size, size_t block_size, ArenaFree free_fn);

    // Allocate n blocks of `size` bytes each; invokes cb once per block.
    // On failure frees any partial allocations and returns false.
    bool allocate(size_t size, size_t n, const AllocationCallback& cb);
    bool deallocate(void* ptr, size_t size, int pool_idx);

    bool need_extend() const;
    size_t total_blocks() const;
    size_t used_blocks() const;
    size_t num_pools() const { return pools_.size(); }
    const MemoryPool* pool(int idx) const {
        return idx >= 0 && static_cast<size_t>(idx) < pools_.size() ? pools_[idx].get() : nullptr;
    }
    MemoryPool* pool_mut(int idx) {
        return idx >= 0 && static_cast<size_t>(idx) < pools_.size() ? pools_[idx].get() : nullptr;
    }

   private:
    std::vector<std::unique_ptr<MemoryPool>> pools_;
    std::vector<ArenaFree> free_fns_;
};

}  // namespace ifs
