#include "mempool.h"

#include <algorithm>
#include <cassert>

#include "log.h"

namespace ifs {

MemoryPool::MemoryPool(void* base, size_t size, size_t block_size, int pool_idx)
    : base_(base), size_(size), block_size_(block_size), pool_idx_(pool_idx) {
    n_blocks_ = size / block_size;
    n_words_ = (n_blocks_ + 63) / 64;
    bits_.assign(n_words_, 0);
    // Mark the tail of the last partial word as used so it is never handed out.
    size_t tail = n_words_ * 64 - n_blocks_;
    if (tail) bits_[n_words_ - 1] = ~0ull << (64 - tail);
    summary_.assign((n_words_ + 63) / 64, 0);
}

bool MemoryPool::run_is_free(size_t start, size_t nb) const {
    if (start + nb > n_blocks_) return false;
    size_t w = start / 64, b = start % 64;
    size_t left = nb;
    while (left) {
        size_t take = std::min<size_t>(64 - b, left);
        uint64_t mask = (take == 64) ? ~0ull : (((1ull << take) - 1) << b);
        if (bits_[w] & mask) return false;
        left -= take;
        w++;
        b = 0;
    }
    return true;
}

void MemoryPool::mark(size_t start, size_t nb, bool used) {
    size_t w = start / 64, b = start % 64;
    size_t left = nb;
    while (left) {
        size_t take = std::min<size_t>(64 - b, left);
        uint64_t mask = (take == 64) ? ~0ull : (((1ull << take) - 1) << b);
        if (used)
            bits_[w] |= mask;
        else
            bits_[w] &= ~mask;
        // Maintain the summary bit for this word.
        uint64_t full = bits_[w] == ~0ull;
        uint64_t sbit = 1ull << (w % 64);
        if (full)
            summary_[w / 64] |= sbit;
        else
            summary_[w / 64] &= ~sbit;
        left -= take;
        w++;
        b = 0;
    }
}

bool MemoryPool::find_run(size_t nb, size_t* out_start) {
    // Two passes: from cursor to end, then from 0 to cursor.
    if (find_run_in(nb, cursor_, n_words_, n_blocks_, out_start)) {
        cursor_ = *out_start / 64;
        return true;
    }
    if (find_run_in(nb, 0, cursor_, n_blocks_, out_start)) {
        cursor_ = *out_start / 64;
        return true;
    }
    return false;
}

bool MemoryPool::find_run_in(size_t nb, size_t w_begin, size_t w_end, size_t block_limit,
                             size_t* out_start) {
    {
        size_t run_start = 0, run_len = 0;
        for (size_t w = w_begin; w < w_end;) {
            // Skip fully-used words fast via the summary (only when we are
            // not extending a run across a boundary — a full word breaks any
            // run anyway, so resetting is correct).
            if (bits_[w] == ~0ull) {
                run_len = 0;
                w++;
                // Jump over whole summary words (4096 fully-used blocks each).
                while (w + 64 <= w_end && (w % 64) == 0 && summary_[w / 64] == ~0ull) w += 64;
                continue;
            }
            uint64_t word = bits_[w];
            if (word == 0) {
                if (run_len == 0) run_start = w * 64;
                run_len += 64;
                if (run_len >= nb && run_start + nb <= block_limit) {
                    *out_start = run_start;
                    return true;
                }
                w++;
                continue;
            }
            // Mixed word: walk its bits.
            for (size_t b = 0; b < 64; b++) {
                size_t idx = w * 64 + b;
                if (idx >= n_blocks_) break;
                if (word & (1ull << b)) {
                    run_len = 0;
                } else {
                    if (run_len == 0) run_start = idx;
                    run_len++;
                    if (run_len >= nb && run_start + nb <= block_limit) {
                        *out_start = run_start;
                        return true;
                    }
                }
            }
            w++;
        }
    }
    return false;
}

void* MemoryPool::allocate(size_t size) {
    if (size == 0 || size > size_) return nullptr;
    size_t nb = (size + block_size_ - 1) / block_size_;
    size_t start;
    if (nb == 1) {
        // O(1) churn path: pop cached free indices, skipping stale hints.
        while (!free_stack_.empty()) {
            uint32_t idx = free_stack_.back();
            free_stack_.pop_back();
            if (!(bits_[idx / 64] & (1ull << (idx % 64)))) {
                mark(idx, 1, true);
                used_blocks_ += 1;
                return static_cast<uint8_t*>(base_) + size_t(idx) * block_size_;
            }
        }
    }
    if (!find_run(nb, &start)) return nullptr;
    mark(start, nb, true);
    used_blocks_ += nb;
    return static_cast<uint8_t*>(base_) + start * block_size_;
}

void* MemoryPool::allocate_contiguous(size_t size, size_t n) {
    if (size == 0 || n == 0) return nullptr;
    size_t nb = (size + block_size_ - 1) / block_size_;
    size_t total = nb * n;
    if (total > n_blocks_ || total > n_blocks_ - used_blocks_) return nullptr;
    if (total > 1 && !contig_ok_) return nullptr;  // fragmented: don't rescan
    size_t start;
    if (!find_run(total, &start)) {
        if (total > 1) {
            contig_ok_ = false;  // re-armed after enough frees (deallocate)
            frees_since_contig_fail_ = 0;
        }
        return nullptr;
    }
    mark(start, total, true);
    used_blocks_ += total;
    return static_cast<uint8_t*>(base_) + start * block_size_;
}

void* MemoryPool::allocate_below(size_t size, size_t limit_block) {
    if (size == 0 || size > size_) return nullptr;
    size_t nb = (size + block_size_ - 1) / block_size_;
    size_t start;
    size_t w_end = std::min(n_words_, (limit_block + 63) / 64);
    if (!find_run_in(nb, 0, w_end, limit_block, &start)) return nullptr;
    mark(start, nb, true);
    used_blocks_ += nb;
    return static_cast<uint8_t*>(base_) + start * block_size_;
}

size_t MemoryPool::largest_free_run() const {
    size_t best = 0, run = 0;
    for (size_t i = 0; i < n_blocks_; i++) {
        if (bits_[i / 64] & (1ull << (i % 64))) {
            run = 0;
        } else {
            run++;
            if (run > best) best = run;
        }
    }
    return best;
}

bool MemoryPool::deallocate(void* ptr, size_t size) {
    auto p = reinterpret_cast<uintptr_t>(ptr);
    auto b = reinterpret_cast<uintptr_t>(base_);
    if (p < b || p >= b + size_ || (p - b) % block_size_ != 0) {
        ERROR("deallocate: invalid pointer %p for pool %d", ptr, pool_idx_);
        return false;
    }
    size_t start = (p - b) / block_size_;
    size_t nb = (size + block_size_ - 1) / block_size_;
    if (start + nb > n_blocks_) return false;
    // Double-free detection: every block in the run must be marked used.
    size_t w = start / 64, bit = start % 64, left = nb;
    while (left) {
        size_t take = std::min<size_t>(64 - bit, left);
        uint64_t mask = (take == 64) ? ~0ull : (((1ull << take) - 1) << bit);
        if ((bits_[w] & mask) != mask) {
            ERROR("deallocate: double free at block %zu in pool %d", start, pool_idx_);
            return false;
        }
        left -= take;
        w++;
        bit = 0;
    }
    mark(start, nb, false);
    used_blocks_ -= nb;
    if (nb == 1 && free_stack_.size() < n_blocks_) {
        free_stack_.push_back(static_cast<uint32_t>(start));
    }
    if (!contig_ok_ && ++frees_since_contig_fail_ >= 4096) contig_ok_ = true;
    return true;
}

MM::~MM() {
    for (size_t i = 0; i < pools_.size(); i++) {
        if (free_fns_[i]) free_fns_[i](pools_[i]->base(), pools_[i]->size());
    }
}

int MM::add_pool(void* base, size_t size, size_t block_size, ArenaFree free_fn) {
    int idx = static_cast<int>(pools_.size());
    pools_.emplace_back(new MemoryPool(base, size, block_size, idx));
    free_fns_.push_back(std::move(free_fn));
    INFO("pool %d added: %zu MB, block %zu KB, %zu blocks", idx, size >> 20, block_size >> 10,
         pools_.back()->total_blocks());
    return idx;
}

bool MM::allocate(size_t size, size_t n, const AllocationCallback& cb) {
    // Batch fast path: one contiguous run for the whole request.
    if (n > 1) {
        for (auto& pool : pools_) {
            void* base = pool->allocate_contiguous(size, n);
            if (base) {
                size_t stride =
                    ((size + pool->block_size() - 1) / pool->block_size()) * pool->block_size();
                for (size_t i = 0; i < n; i++)
                    cb(static_cast<uint8_t*>(base) + i * stride, pool->pool_idx());
                return true;
            }
        }
        // fall through to per-page allocation across pools
    }
    struct Undo {
        void* ptr;
        int pool;
    };
    std::vector<Undo> done;
    done.reserve(n);
    for (size_t i = 0; i < n; i++) {
        void* p = nullptr;
        int pidx = -1;
        for (auto& pool : pools_) {
            p = pool->allocate(size);
            if (p) {
                pidx = pool->pool_idx();
                break;
            }
        }
        if (!p) {
            for (auto& u : done) pools_[u.pool]->deallocate(u.ptr, size);
            WARN("allocation failed: %zu x %zu bytes (used %zu/%zu blocks)", n, size,
                 used_blocks(), total_blocks());
            return false;
        }
        done.push_back({p, pidx});
    }
    for (auto& u : done) cb(u.ptr, u.pool);
    return true;
}

bool MM::deallocate(void* ptr, size_t size, int pool_idx) {
    if (pool_idx < 0 || static_cast<size_t>(pool_idx) >= pools_.size()) return false;
    return pools_[pool_idx]->deallocate(ptr, size);
}

bool MM::need_extend() const {
    if (pools_.empty()) return false;
    auto& last = pools_.back();
    return last->used_blocks() > last->total_blocks() * kBlockUsageRatio;
}

size_t MM::total_blocks() const {
    size_t t = 0;
    for (auto& p : pools_) t += p->total_blocks();
    return t;
}

size_t MM::used_blocks() const {
    size_t t = 0;
    for (auto& p : pools_) t += p->used_blocks();
    return t;
}

}  // namespace ifs
