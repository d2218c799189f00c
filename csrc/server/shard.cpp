#include "shard.h"

#include <algorithm>
#include <cstdlib>
#include <cstring>

#include "../core/log.h"

namespace ifs {

Shard::Shard(const ShardOptions& opt) : opt_(opt) {}

Shard::~Shard() {
    for (auto& sc : streams_) {
        std::lock_guard<std::mutex> lk(sc.mu);
        stopping_ = true;
    }
    stopping_ = true;  // also when there are no streams (CPU shard)
    for (auto& sc : streams_) {
        sc.task_cv.notify_all();
        sc.slot_cv.notify_all();
    }
    {
        std::lock_guard<std::mutex> lk(fabric_mu_);
    }
    fabric_cv_.notify_all();
    if (fabric_thread_.joinable()) fabric_thread_.join();
    for (auto& t : completion_threads_)
        if (t.joinable()) t.join();
    for (auto& sc : streams_) {
        for (auto& sl : sc.slots) {
            if (sl.event) gpu::event_destroy(sl.event);
            if (sl.h_src) gpu::free_host_pinned(sl.h_src);
            if (sl.h_dst) gpu::free_host_pinned(sl.h_dst);
            if (sl.h_scale) gpu::free_host_pinned(sl.h_scale);
        }
        if (sc.stream) gpu::stream_destroy(sc.stream);
    }
    // MM frees arenas via its free_fns.
}

bool Shard::init() {
    if (inited_) return true;
    size_t desc_bytes = opt_.max_descs_per_slot * sizeof(uint64_t);
    if (on_gpu()) {
        void* arena = gpu::alloc_device(opt_.device, opt_.pool_bytes);
        if (!arena) {
            ERROR("shard %d: hipMalloc %zu MB failed", opt_.device, opt_.pool_bytes >> 20);
            return false;
        }
        {
            std::lock_guard<std::mutex> lk(alloc_mu_);
            mm_.add_pool(arena, opt_.pool_bytes, opt_.block_granule,
                         [](void* p, size_t) { gpu::free_device(p); });
        }
        streams_.resize(opt_.n_streams);
        for (auto& sc : streams_) {
            sc.stream = gpu::stream_create(opt_.device);
            if (!sc.stream) return false;
            sc.slots.resize(opt_.slots_per_stream);
            for (auto& sl : sc.slots) {
                size_t scale_bytes = opt_.max_descs_per_slot * sizeof(float);
                sl.h_src = static_cast<uint64_t*>(gpu::alloc_host_pinned(desc_bytes));
                sl.h_dst = static_cast<uint64_t*>(gpu::alloc_host_pinned(desc_bytes));
                sl.h_scale = static_cast<float*>(gpu::alloc_host_pinned(scale_bytes));
                sl.event = gpu::event_create(opt_.device);
                if (!sl.h_src || !sl.h_dst || !sl.h_scale || !sl.event) return false;
            }
        }
        // One completion thread PER STREAM: done-callbacks can block (the
        // shm write commit waits for the insert pass) and a single thread
        // serializing all streams would convoy an already-finished read's
        // reply behind a blocked write commit.
        for (size_t si = 0; si < streams_.size(); si++)
            completion_threads_.emplace_back([this, si] { completion_loop(si); });
    } else {
        void* arena = nullptr;
        if (posix_memalign(&arena, 4096, opt_.pool_bytes) != 0) {
            ERROR("cpu shard: alloc %zu MB failed", opt_.pool_bytes >> 20);
            return false;
        }
        std::lock_guard<std::mutex> lk(alloc_mu_);
        mm_.add_pool(arena, opt_.pool_bytes, opt_.block_granule, [](void* p, size_t) { free(p); });
    }
    fabric_thread_ = std::thread([this] { fabric_loop(); });
    inited_ = true;
    return true;
}

bool Shard::allocate(size_t size, size_t n, const AllocationCallback& cb) {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    return mm_.allocate(size, n, cb);
}

void Shard::deallocate_bulk(const std::vector<BlockFree>& frees) {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    for (auto& f : frees) mm_.deallocate(f.ptr, f.size, f.pool_idx);
}

bool Shard::deallocate(void* ptr, size_t size, int pool_idx) {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    return mm_.deallocate(ptr, size, pool_idx);
}

size_t Shard::used_blocks() {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    return mm_.used_blocks();
}

size_t Shard::total_blocks() {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    return mm_.total_blocks();
}

bool Shard::contains(const void* p) {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    for (size_t i = 0; i < mm_.num_pools(); i++)
        if (mm_.pool(static_cast<int>(i))->contains(const_cast<void*>(p))) return true;
    return false;
}

bool Shard::need_extend() {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    return opt_.auto_extend && mm_.need_extend();
}

bool Shard::extend(void** arena_out) {
    if (on_gpu()) {
        void* arena = gpu::alloc_device(opt_.device, opt_.extend_bytes);
        if (!arena) return false;
        std::lock_guard<std::mutex> lk(alloc_mu_);
        mm_.add_pool(arena, opt_.extend_bytes, opt_.block_granule,
                     [](void* p, size_t) { gpu::free_device(p); });
        if (arena_out) *arena_out = arena;
    } else {
        void* arena = nullptr;
        if (posix_memalign(&arena, 4096, opt_.extend_bytes) != 0) return false;
        std::lock_guard<std::mutex> lk(alloc_mu_);
        mm_.add_pool(arena, opt_.extend_bytes, opt_.block_granule,
                     [](void* p, size_t) { free(p); });
        if (arena_out) *arena_out = arena;
    }
    return true;
}

std::vector<Shard::Move> Shard::plan_compaction(
    const std::vector<std::pair<void*, size_t>>& movable) {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    // Move highest addresses first so freed tail space cascades.
    auto sorted = movable;
    std::sort(sorted.begin(), sorted.end(),
              [](const auto& a, const auto& b) { return a.first > b.first; });
    std::vector<Move> moves;
    for (auto& [ptr, size] : sorted) {
        MemoryPool* owner = nullptr;
        for (size_t i = 0; i < mm_.num_pools(); i++) {
            MemoryPool* p = mm_.pool_mut(static_cast<int>(i));
            if (p->contains(ptr)) {
                owner = p;
                break;
            }
        }
        if (!owner) continue;
        size_t idx = owner->block_index_of(ptr);
        void* np = owner->allocate_below(size, idx);
        if (np) moves.push_back({ptr, np, size, owner->pool_idx()});
    }
    return moves;
}

void Shard::for_each_arena(const std::function<void(void*, size_t, bool)>& fn) {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    for (size_t i = 0; i < mm_.num_pools(); i++) {
        const MemoryPool* p = mm_.pool(static_cast<int>(i));
        fn(p->base(), p->size(), on_gpu());
    }
}

size_t Shard::largest_free_run_bytes() {
    std::lock_guard<std::mutex> lk(alloc_mu_);
    size_t best = 0;
    for (size_t i = 0; i < mm_.num_pools(); i++) {
        const MemoryPool* p = mm_.pool(static_cast<int>(i));
        best = std::max(best, p->largest_free_run() * p->block_size());
    }
    return best;
}

Shard::Slot* Shard::acquire_slot(StreamCtx& sc) {
    static const bool dbg = getenv("IFS_SERVER_DEBUG") != nullptr;
    auto t0 = dbg ? std::chrono::steady_clock::now() : std::chrono::steady_clock::time_point{};
    std::unique_lock<std::mutex> lk(sc.mu);
    for (;;) {
        int n = static_cast<int>(sc.slots.size());
        for (int i = 0; i < n; i++) {
            Slot& sl = sc.slots[(sc.next_slot + i) % n];
            if (!sl.busy) {
                sl.busy = true;
                sc.next_slot = (sc.next_slot + i + 1) % n;
                if (dbg) {
                    auto us = std::chrono::duration<double, std::micro>(
                                  std::chrono::steady_clock::now() - t0)
                                  .count();
                    if (us > 1000)
                        fprintf(stderr, "[sdbg] slot wait %.0fus (stream backlog)\n", us);
                }
                return &sl;
            }
        }
        sc.slot_cv.wait(lk);
        if (stopping_) return nullptr;
    }
}

bool Shard::submit_copy(CopyJob&& job) {
    size_t n = job.src.size();
    if (n != job.dst.size() || job.bytes_per_block == 0) return false;
    if (n == 0) {
        if (job.done) job.done(true);
        return true;
    }

    // Fan very large jobs out across the stream pool: one 1 GB request as a
    // single kernel leaves other streams idle and interacts badly with
    // concurrent large requests; ~4096-block (512 MB at 128 KB) sub-jobs
    // spread the work and complete independently.
    constexpr size_t kFanChunk = 4096;
    if (on_gpu() && n > kFanChunk && streams_.size() > 1) {
        size_t n_sub = (n + kFanChunk - 1) / kFanChunk;
        auto pending = std::make_shared<std::atomic<int>>(static_cast<int>(n_sub));
        auto all_ok = std::make_shared<std::atomic<bool>>(true);
        auto done = std::make_shared<std::function<void(bool)>>(std::move(job.done));
        for (size_t off = 0; off < n; off += kFanChunk) {
            size_t take = std::min(kFanChunk, n - off);
            CopyJob sub;
            sub.bytes_per_block = job.bytes_per_block;
            sub.xform = job.xform;
            sub.scales_out = job.scales_out;
            sub.scales_off = job.scales_off + off;
            if (!job.scales_in.empty())
                sub.scales_in.assign(job.scales_in.begin() + static_cast<long>(off),
                                     job.scales_in.begin() + static_cast<long>(off + take));
            sub.src.assign(job.src.begin() + static_cast<long>(off),
                           job.src.begin() + static_cast<long>(off + take));
            sub.dst.assign(job.dst.begin() + static_cast<long>(off),
                           job.dst.begin() + static_cast<long>(off + take));
            sub.done = [pending, all_ok, done](bool ok) {
                if (!ok) all_ok->store(false);
                if (pending->fetch_sub(1) == 1 && *done) (*done)(all_ok->load());
            };
            if (!submit_copy(std::move(sub))) {
                all_ok->store(false);
                if (pending->fetch_sub(1) == 1 && *done) (*done)(all_ok->load());
            }
        }
        return true;
    }

    if (!on_gpu()) {
        if (job.xform != CopyJob::Xform::kCopy) return false;  // GPU-only
        // CPU shard: copies run inline on the caller (loop) thread.
        for (size_t i = 0; i < n; i++)
            memcpy(reinterpret_cast<void*>(job.dst[i]), reinterpret_cast<const void*>(job.src[i]),
                   job.bytes_per_block);
        if (job.done) job.done(true);
        return true;
    }

    // Alignment check for the vectorized kernel.
    bool aligned = job.bytes_per_block % 16 == 0;
    if (aligned) {
        for (size_t i = 0; i < n && aligned; i++)
            aligned = (job.src[i] % 16 == 0) && (job.dst[i] % 16 == 0);
    }

    // HIP stream enqueues are thread-safe; each job's ops precede its event
    // in stream order regardless of interleaving with other submitters, so
    // no submit-wide lock is needed (a lock here convoyed all clients behind
    // slot waits). Stream choice: round-robin, but skip ahead to a stream
    // with a free slot when the nominal one is full (a job's chunks must
    // stay on ONE stream — completion order per stream is what fires the
    // done callback after the last chunk).
    uint32_t rr = next_stream_.fetch_add(1);
    uint32_t n_streams = static_cast<uint32_t>(streams_.size());
    uint32_t pick = rr % n_streams;
    auto has_free_slot = [](StreamCtx& cand) {
        std::lock_guard<std::mutex> lk(cand.mu);
        for (auto& sl : cand.slots)
            if (!sl.busy) return true;
        return false;
    };
    // Round-robin keeps concurrent writers/readers spread across streams
    // (their kernels overlap); only when the nominal stream is FULL look
    // for another with room. (An always-scan variant made every submitter
    // converge on whichever stream had just freed slots, collapsing the
    // pipelined few-connection bench onto one stream.)
    if (!has_free_slot(streams_[pick])) {
        for (uint32_t i = 1; i < n_streams; i++) {
            if (has_free_slot(streams_[(rr + i) % n_streams])) {
                pick = (rr + i) % n_streams;
                break;
            }
        }
    }
    StreamCtx& sc = streams_[pick];

    // Small aligned batches: descriptors ride in the kernel arguments.
    if (aligned && n <= 16 && job.xform == CopyJob::Xform::kCopy) {
        Slot* slot = acquire_slot(sc);
        if (!slot) return false;
        bool ok = gpu::set_device(opt_.device) &&
                  gpu::launch_copy_blocks_inline(opt_.device, sc.stream, job.src.data(),
                                                 job.dst.data(), static_cast<int>(n),
                                                 job.bytes_per_block) &&
                  gpu::event_record(slot->event, sc.stream);
        {
            std::lock_guard<std::mutex> lk(sc.mu);
            if (!ok) slot->busy = false;
            sc.pending.push_back({ok ? slot : nullptr, std::move(job.done), nullptr, 0, 0});
        }
        sc.task_cv.notify_one();
        return true;
    }

    // Chunk over slot capacity; the done callback fires after the final chunk
    // (chunks on one stream complete in order).
    size_t cap = opt_.max_descs_per_slot;
    size_t off = 0;
    while (off < n) {
        size_t take = std::min(cap, n - off);
        Slot* slot = acquire_slot(sc);
        if (!slot) return false;  // shutting down
        memcpy(slot->h_src, job.src.data() + off, take * sizeof(uint64_t));
        memcpy(slot->h_dst, job.dst.data() + off, take * sizeof(uint64_t));
        bool last = off + take >= n;
        bool ok = gpu::set_device(opt_.device);
        // Descriptors live in pinned host memory for EVERY size: the copy
        // kernels read each block's descriptor once per workgroup (LDS
        // broadcast), so there is no per-unit PCIe tax and no per-job
        // upload (whose rocclr blit cost ~110 µs,
        // profiles/rocprof_bench_r02.txt).
        const uint64_t* dsrc = slot->h_src;
        const uint64_t* ddst = slot->h_dst;
        if (job.xform == CopyJob::Xform::kCopy) {
            ok = ok && gpu::launch_copy_blocks(opt_.device, sc.stream, dsrc, ddst,
                                               static_cast<int>(take), job.bytes_per_block,
                                               aligned);
        } else if (job.xform == CopyJob::Xform::kQuantBf16Fp8) {
            // The kernel writes each block's scale straight into pinned
            // h_scale (one 4 B write per block) — no D2H needed.
            ok = ok && gpu::launch_quant_blocks(opt_.device, sc.stream, dsrc, ddst,
                                                slot->h_scale, static_cast<int>(take),
                                                job.bytes_per_block / 2);
        } else {  // kDequantFp8Bf16 (kernel reads desc + scale once per WG)
            memcpy(slot->h_scale, job.scales_in.data() + off, take * sizeof(float));
            ok = ok && gpu::launch_dequant_blocks(opt_.device, sc.stream, dsrc, ddst,
                                                  slot->h_scale, static_cast<int>(take),
                                                  job.bytes_per_block / 2);
        }
        ok = ok && gpu::event_record(slot->event, sc.stream);
        if (!ok) {
            // Report the failure through the done callback (even if earlier
            // chunks were already queued) and stop submitting.
            ERROR("shard %d submit failed: %s", opt_.device, gpu::last_error());
            {
                std::lock_guard<std::mutex> lk(sc.mu);
                slot->busy = false;
                sc.pending.push_back({nullptr, std::move(job.done), nullptr, 0, 0});
            }
            sc.slot_cv.notify_one();
            sc.task_cv.notify_one();
            return true;
        }
        {
            std::lock_guard<std::mutex> lk(sc.mu);
            bool q = job.xform == CopyJob::Xform::kQuantBf16Fp8;
            sc.pending.push_back({slot,
                                  last ? std::move(job.done) : std::function<void(bool)>(),
                                  q ? job.scales_out : nullptr, job.scales_off + off, take});
        }
        sc.task_cv.notify_one();
        off += take;
    }
    return true;
}

void Shard::completion_loop(size_t stream_idx) {
    gpu::set_device(opt_.device);
    // One thread per stream, so streams complete fully independently: a
    // done-callback that blocks (e.g. the shm write commit waiting for the
    // insert pass) stalls only its own stream's FIFO.
    auto& sc = streams_[stream_idx];
    std::vector<PendingTask> batch;
    for (;;) {
        batch.clear();
        {
            std::unique_lock<std::mutex> lk(sc.mu);
            for (;;) {
                if (stopping_ && sc.pending.empty()) return;
                if (sc.pending.empty()) {
                    sc.task_cv.wait(lk, [&] { return stopping_ || !sc.pending.empty(); });
                    continue;
                }
                // Events on one stream complete in order, so if pending[k]'s
                // event fired, 0..k all did: binary-search the completed
                // prefix and take it in ONE pass. This keeps hipEventQuery
                // traffic at O(log depth) per sweep instead of hammering the
                // HIP runtime's locks from a per-task poll (which competed
                // with the 64 handler threads' kernel launches).
                size_t n = sc.pending.size();
                size_t done_n = 0;
                // Null-slot tasks (submit failures) are trivially complete.
                while (done_n < n && !sc.pending[done_n].slot) done_n++;
                if (done_n < n) {
                    size_t lo = done_n, hi = n;  // invariant: [0,lo) complete
                    while (lo < hi) {
                        size_t mid = (lo + hi) / 2;
                        if (!sc.pending[mid].slot ||
                            gpu::event_query(sc.pending[mid].slot->event))
                            lo = mid + 1;
                        else
                            hi = mid;
                    }
                    done_n = lo;
                }
                if (done_n > 0) {
                    for (size_t i = 0; i < done_n; i++) {
                        batch.push_back(std::move(sc.pending.front()));
                        sc.pending.pop_front();
                    }
                    break;
                }
                // Nothing complete yet: poll the front event off-lock with
                // a short backoff (hipEventSynchronize's interrupt-path
                // wake costs ~100 µs per event and serialized slot drains).
                gpu::Event ev = sc.pending.front().slot->event;
                lk.unlock();
                // Spin through the expected completion window (~hundreds of
                // µs for a batched copy) before sleeping: usleep's real
                // granularity is ~60 µs, and paying it per completion put
                // ~15% on the pipelined bench and a 150 µs p99 on the
                // single-block round trip.
                auto tq0 = std::chrono::steady_clock::now();
                while (!gpu::event_query(ev) && !stopping_) {
#if defined(__x86_64__)
                    for (int i = 0; i < 64; i++) __builtin_ia32_pause();
#endif
                    auto us = std::chrono::duration<double, std::micro>(
                                  std::chrono::steady_clock::now() - tq0)
                                  .count();
                    if (us > 2000) usleep(50);  // ms-scale kernel: back off
                }
                lk.lock();
            }
        }
        // Copy out fp8 scales and release every slot in one lock hold, then
        // run the done callbacks in order outside the lock.
        {
            std::lock_guard<std::mutex> lk(sc.mu);
            for (auto& t : batch) {
                if (t.slot) {
                    if (t.scales_out && t.scales_n)
                        memcpy(t.scales_out->data() + t.scales_off, t.slot->h_scale,
                               t.scales_n * sizeof(float));
                    t.slot->busy = false;
                }
            }
        }
        if (batch.size() > 1)
            sc.slot_cv.notify_all();
        else
            sc.slot_cv.notify_one();
        for (auto& t : batch) {
            if (t.done) t.done(t.slot != nullptr);
        }
    }
}


void Shard::submit_fabric(FabricJob&& job) {
    {
        std::lock_guard<std::mutex> lk(fabric_mu_);
        fabric_q_.push_back(std::move(job));
    }
    fabric_cv_.notify_one();
}

bool Shard::fabric_stage_init() {
    if (fab_ready_) return true;
    fab_stream_ = gpu::stream_create(opt_.device);
    if (!fab_stream_) return false;
    for (auto& b : fab_) {
        b.h_stage = static_cast<uint8_t*>(gpu::alloc_host_pinned(kFabStageBytes));
        b.d_stage = static_cast<uint8_t*>(gpu::alloc_device(opt_.device, kFabStageBytes));
        b.h_desc = static_cast<uint64_t*>(
            gpu::alloc_host_pinned(2 * kFabStageDescs * sizeof(uint64_t)));
        b.d_desc = static_cast<uint64_t*>(
            gpu::alloc_device(opt_.device, 2 * kFabStageDescs * sizeof(uint64_t)));
        b.event = gpu::event_create(opt_.device);
        if (!b.h_stage || !b.d_stage || !b.h_desc || !b.d_desc || !b.event) {
            fabric_stage_teardown();
            return false;
        }
    }
    fab_ready_ = true;
    return true;
}

void Shard::fabric_stage_teardown() {
    for (auto& b : fab_) {
        if (b.h_stage) gpu::free_host_pinned(b.h_stage);
        if (b.d_stage) gpu::free_device(b.d_stage);
        if (b.h_desc) gpu::free_host_pinned(b.h_desc);
        if (b.d_desc) gpu::free_device(b.d_desc);
        if (b.event) gpu::event_destroy(b.event);
        b = FabBuf{};
    }
    if (fab_stream_) gpu::stream_destroy(fab_stream_);
    fab_stream_ = nullptr;
    fab_ready_ = false;
}

// Staged GPU fabric transfer. PUT: pageable body -> pinned stage (memcpy),
// one H2D of the chunk, one scatter-kernel launch stage->pool blocks.
// GET: one gather-kernel launch pool->stage, one D2H, memcpy out into the
// response buffer. Chunks alternate between two buffers; the host memcpy
// of one chunk overlaps the device work of the other, so the sustained
// rate is bounded by host memcpy bandwidth, not PCIe round trips.
bool Shard::fabric_run_gpu(FabricJob& job, uint8_t* base) {
    const size_t bs = job.bytes_per_block;
    const size_t n = job.block_ptrs.size();
    const size_t per_chunk = std::min(kFabStageBytes / bs, kFabStageDescs);
    if (per_chunk == 0) return false;  // block larger than the stage: refuse
    const bool aligned = bs % 16 == 0;
    bool ok = true;

    auto flush = [&](FabBuf& b) {  // wait chunk; for GET copy payload out
        if (!b.in_flight) return;
        if (ok) ok = gpu::event_sync(b.event);
        if (ok && !job.is_put) {
            for (size_t i = 0; i < b.out_n; i++)
                memcpy(base + job.host_offsets[b.out_first + i], b.h_stage + i * bs, bs);
        }
        b.in_flight = false;
    };

    int cur = 0;
    for (size_t off = 0; off < n && ok; off += per_chunk) {
        FabBuf& b = fab_[cur];
        flush(b);  // buffer must be idle before reuse
        if (!ok) break;
        const size_t take = std::min(per_chunk, n - off);
        if (job.is_put) {
            for (size_t i = 0; i < take; i++)
                memcpy(b.h_stage + i * bs, base + job.host_offsets[off + i], bs);
            for (size_t i = 0; i < take; i++) {
                b.h_desc[i] = reinterpret_cast<uint64_t>(b.d_stage + i * bs);
                b.h_desc[kFabStageDescs + i] = job.block_ptrs[off + i];
            }
            ok = gpu::memcpy_h2d_async(b.d_stage, b.h_stage, take * bs, fab_stream_) &&
                 gpu::memcpy_h2d_async(b.d_desc, b.h_desc, take * sizeof(uint64_t),
                                       fab_stream_) &&
                 gpu::memcpy_h2d_async(b.d_desc + kFabStageDescs,
                                       b.h_desc + kFabStageDescs, take * sizeof(uint64_t),
                                       fab_stream_) &&
                 gpu::launch_copy_blocks(opt_.device, fab_stream_, b.d_desc,
                                         b.d_desc + kFabStageDescs, static_cast<int>(take),
                                         bs, aligned) &&
                 gpu::event_record(b.event, fab_stream_);
        } else {
            for (size_t i = 0; i < take; i++) {
                b.h_desc[i] = job.block_ptrs[off + i];
                b.h_desc[kFabStageDescs + i] = reinterpret_cast<uint64_t>(b.d_stage + i * bs);
            }
            b.out_first = off;
            b.out_n = take;
            ok = gpu::memcpy_h2d_async(b.d_desc, b.h_desc, take * sizeof(uint64_t),
                                       fab_stream_) &&
                 gpu::memcpy_h2d_async(b.d_desc + kFabStageDescs,
                                       b.h_desc + kFabStageDescs, take * sizeof(uint64_t),
                                       fab_stream_) &&
                 gpu::launch_copy_blocks(opt_.device, fab_stream_, b.d_desc,
                                         b.d_desc + kFabStageDescs, static_cast<int>(take),
                                         bs, aligned) &&
                 gpu::memcpy_d2h_async(b.h_stage, b.d_stage, take * bs, fab_stream_) &&
                 gpu::event_record(b.event, fab_stream_);
        }
        b.in_flight = ok;
        cur ^= 1;
    }
    flush(fab_[cur]);
    flush(fab_[cur ^ 1]);
    return ok;
}

void Shard::fabric_loop() {
    if (on_gpu()) gpu::set_device(opt_.device);
    for (;;) {
        FabricJob job;
        {
            std::unique_lock<std::mutex> lk(fabric_mu_);
            fabric_cv_.wait(lk, [this] { return stopping_ || !fabric_q_.empty(); });
            if (stopping_ && fabric_q_.empty()) break;
            if (fabric_q_.empty()) continue;
            job = std::move(fabric_q_.front());
            fabric_q_.pop_front();
        }
        uint8_t* base = job.raw_host ? job.raw_host.get()
                                     : (job.host ? job.host->data() : nullptr);
        bool ok = base != nullptr;
        size_t bs = job.bytes_per_block;
        if (ok && on_gpu() && bs > 0 && bs <= kFabStageBytes && fabric_stage_init()) {
            ok = fabric_run_gpu(job, base);
        } else if (ok) {
            // CPU shard (or staging alloc failure): plain memcpy per block.
            for (size_t i = 0; i < job.block_ptrs.size() && ok; i++) {
                uint8_t* hp = base + job.host_offsets[i];
                void* bp = reinterpret_cast<void*>(job.block_ptrs[i]);
                if (on_gpu())
                    ok = job.is_put ? gpu::memcpy_h2d(bp, hp, bs) : gpu::memcpy_d2h(hp, bp, bs);
                else if (job.is_put)
                    memcpy(bp, hp, bs);
                else
                    memcpy(hp, bp, bs);
            }
        }
        if (job.done) job.done(ok);
    }
    if (on_gpu()) fabric_stage_teardown();
}

}  // namespace ifs
