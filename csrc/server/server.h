// infinistore-amd server core.
//
// Same externally-visible behavior as the reference server
// (/root/reference/src/infinistore.cpp): 9-byte-header TCP protocol, string
// keys -> fixed-size committed/uncommitted blocks, dedup on write, purge,
// prefix match — but a new MI355X-native architecture:
//
//  * The libuv loop runs on a DEDICATED C++ thread owned by this class; the
//    Python process keeps its own asyncio loop for the management plane
//    (the reference instead threads uvloop's uv_loop_t* through a PyCapsule
//    into C++, infinistore.cpp:1260, lib.py:193-200 — designed away).
//  * The pool is sharded per MI355X GPU (HBM3E arenas, csrc/server/shard.h);
//    local-path requests are served by ONE batched gather/scatter HIP kernel
//    per request instead of a per-block memcpy loop; completions flow back
//    via per-shard completion threads + uv_async.
//  * IPC mappings are cached per connection instead of open/close per
//    request (infinistore.cpp:712/:671), taking hipIpcOpenMemHandle off the
//    hot path.
//  * A TCP data fabric (OP_TCP_PUT/OP_TCP_GET) implements the RDMA-semantics
//    allocate/write/commit/read flows inline over the socket so the full
//    client API works with no RDMA NIC (BASELINE configs 1-2); an ibverbs
//    fabric can slot in behind the same dispatch when rdma-core is present.
#pragma once

#include <uv.h>

#include <array>
#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <shared_mutex>
#include <string>
#include <string_view>
#include <thread>
#include <unordered_map>
#include <vector>

#include "../core/mempool.h"
#include "../core/protocol.h"
#include "../core/shm_ring.h"
#include "../core/utils.h"
#include "../fabric/verbs_fabric.h"
#include "../gpu/gpu.h"
#include "kvmap.h"
#include "shard.h"

namespace ifs {

struct ServerOptions {
    int service_port = 22345;
    size_t prealloc_bytes = 16ull << 30;  // per shard
    size_t block_granule = 64 << 10;      // minimal_allocate_size
    bool auto_extend = false;
    size_t extend_bytes = 10ull << 30;
    std::vector<int> devices;  // GPU ordinals to shard over; empty => CPU shard(s)
    int cpu_shards = 1;        // CPU-mode shard count (tests the routing path)
    int n_streams = 4;
    // IO worker loops in addition to the accept loop. One loop thread
    // saturates near ~10k heavy requests/s; 8 client ranks at full rate need
    // parallel request handling (SURVEY.md §7 hard part 4). 0 = single loop.
    int io_threads = 3;
    std::string log_level = "warning";
    // verbs fabric (used when rdma-core + an active NIC are present)
    std::string dev_name;
    int ib_port = 1;
    std::string link_type = "Ethernet";
    // Opt-in LRU eviction: when an allocation fails, evict the
    // least-recently-accessed committed idle keys and retry (extension — the
    // reference only fails the write; engines can also evict explicitly via
    // delete_keys).
    bool auto_evict = false;
    // Time-to-live for stored keys, seconds; 0 = keys live until
    // deleted/evicted/purged (extension — production prefix caches expire
    // stale prompts; the reference has no expiry at all).
    int ttl_seconds = 0;
};

class Server;

// One stored block. Refcounted: the kv map holds one ref; in-flight reads
// hold another so purge cannot free memory under an active copy.
// Slab-allocated: prefill writes create thousands of entries per request,
// so class-level new/delete run on a chunked freelist instead of malloc.
struct BlockEntry : RefCounted {
    void* ptr = nullptr;
    size_t size = 0;
    int pool_idx = -1;
    Shard* shard = nullptr;
    bool committed = false;
    // fp8-compressed page (extension): `size` is the STORED byte size
    // (half the logical bf16 page); reads dequantize with `scale`.
    bool fp8 = false;
    float scale = 1.f;
    uint32_t born_sec = 0;  // CLOCK_MONOTONIC seconds at insert (TTL)
    // LRU tick (auto_evict); atomic: bumped under the SHARED kv lock by
    // concurrent readers.
    std::atomic<uint64_t> last_access{0};
    ~BlockEntry() override {
        if (shard && ptr) shard->deallocate(ptr, size, pool_idx);
    }
    static void* operator new(size_t n);
    static void operator delete(void* p);
    // placement forms (class-level operator new hides the global ones)
    static void* operator new(size_t, void* p) { return p; }
    static void operator delete(void*, void*) {}
};

class Server {
   public:
    explicit Server(const ServerOptions& opt);
    ~Server();
    Server(const Server&) = delete;
    Server& operator=(const Server&) = delete;

    bool start();  // spawns the loop thread; returns false if bind failed
    void stop();
    bool running() const { return running_.load(); }

    // Management plane (thread-safe; callable from Python).
    size_t kvmap_len();
    size_t purge();
    std::string stats_json();
    // Defragment the pools: move committed, idle blocks to lower addresses
    // using the batched copy kernel. Returns (moved_blocks, moved_bytes).
    std::pair<size_t, size_t> compact();
    // Warm-restart persistence (extension; the reference cache is purely
    // volatile): dump every committed, unexpired entry (key, metadata,
    // page bytes) to `path` / load such a dump back into the pool.
    // Returns (entries, payload_bytes) or (0,0) + false on IO failure.
    bool snapshot(const std::string& path, std::pair<size_t, size_t>* out);
    bool restore(const std::string& path, std::pair<size_t, size_t>* out);
    int num_shards() const { return static_cast<int>(shards_.size()); }

    struct Conn;  // defined below (file-local helpers + server_verbs use it)

    // How a request arrived: over the socket (default) or via the conn's
    // shared-memory ring. Replies for shm requests are written straight into
    // the response ring — from ANY thread — instead of uv_write on the owner
    // loop, which is what lets completion threads answer without the
    // uv_async hop.
    struct ReqCtx {
        uint64_t seq = 0;
        bool shm = false;
    };

    // Shared-memory ring peer (csrc/core/shm_ring.h): one per conn that
    // completed OP_SHM_SETUP. The poller thread holds a conn ref; the
    // segment stays mapped until ~Conn so late completion-thread replies
    // never touch unmapped memory.
    struct ShmPeer {
        Server* srv = nullptr;
        Conn* c = nullptr;
        shmring::Segment seg;
        std::thread th;
        std::atomic<bool> stop{false};
        std::mutex resp_mu;  // serializes response writers (poller + completions)
        void push_resp(uint64_t seq, int status);
    };

    // Uniform view of a local-path request (flatbuffers or packed fast-path
    // format). Key views point into the request body / parsed message and
    // are only valid during the handling call.
    struct LocalView {
        int32_t device = 0;
        int32_t pid = 0;
        uint64_t base_ptr = 0;
        uint64_t base_offset = 0;
        int64_t block_size = 0;
        uint32_t flags = 0;
        uint32_t alloc_mb = 0;  // client allocation size (MB); 0 = unknown
        const uint8_t* ipc = nullptr;
        size_t ipc_len = 0;
        std::vector<std::pair<std::string_view, uint64_t>> blocks;
    };

    // ---- data-plane helpers shared by the TCP and verbs fabrics ----
    // Allocate pages for keys (dedup -> FAKE blocks); fills *status on error.
    std::vector<RemoteBlockWire> allocate_blocks(Conn* c,
                                                 const std::vector<std::string>& keys,
                                                 size_t page, int* status);
    void commit_addrs(Conn* c, const std::vector<uint64_t>& addrs);
    // Collect committed entries for keys; false => missing/uncommitted key.
    bool collect_read_entries(const std::vector<std::string>& keys,
                              std::vector<Ref<BlockEntry>>* out);

    // ---- verbs fabric seam (server_verbs.cpp) ----
    struct VerbsPeer;
    // Attempt a verbs handshake for OP_RDMA_EXCHANGE; fills the reply
    // payload ("VRBS" + ConnInfo) on success.
    bool verbs_handshake(Conn* c, const std::vector<uint8_t>& body,
                         std::vector<uint8_t>* reply);
    void verbs_teardown(Conn* c);  // owner-loop thread; stops the CQ poll

    // One libuv loop + thread. The main loop owns the listener; accepted
    // connections are handed (by fd) round-robin to worker loops so request
    // parsing/dispatch scales across cores. All cross-loop work goes through
    // post().
    struct IoLoop {
        uv_loop_t loop;
        uv_async_t post_async;
        uv_async_t stop_async;
        std::thread thread;
        std::mutex post_mu;
        std::vector<std::function<void()>> posted;
        Server* srv = nullptr;
        std::vector<Conn*> conns;  // touched only from this loop's thread
        bool is_main = false;

        void post(std::function<void()> fn);
        void start();
        void request_stop() { uv_async_send(&stop_async); }
        static void on_post(uv_async_t* h);
        static void on_stop(uv_async_t* h);
    };

   private:
    friend struct VerbsPeer;

    static void on_new_connection(uv_stream_t* server, int status);
    void adopt_fd(IoLoop* io, int fd, bool is_pipe);  // runs on io's thread
    static void on_new_pipe_connection(uv_stream_t* server, int status);
    void post(std::function<void()> fn) { main_io_.post(std::move(fn)); }

    // ---- request handling (loop thread) ----
    void handle_request(Conn* c, char op, std::vector<uint8_t> body);
    void op_local_write(Conn* c, const LocalView& msg, const ReqCtx& ctx);
    void op_local_read(Conn* c, const LocalView& msg, const ReqCtx& ctx);
    void op_sync(Conn* c, const ReqCtx& ctx);
    // remain-- (+ deferred sync reply). on_owner: caller runs on the conn's
    // owner loop thread (socket replies must; shm replies need not).
    void finish_task(Conn* c, bool on_owner);
    // Route a status to the request's transport (ring: any thread; socket:
    // owner loop thread only).
    void reply_local(Conn* c, const ReqCtx& ctx, int code);
    void reply_query(Conn* c, const ReqCtx& ctx, int value);
    void op_shm_setup(Conn* c, const std::vector<uint8_t>& body);
    void shm_teardown(Conn* c);       // owner loop thread; joins the poller
    void shm_poll_main(ShmPeer* p);   // poller thread body
    std::atomic<int> shm_peers_{0};

    // Fast-op worker pool: ring pollers hand OP_W_FAST / OP_R_FAST bodies
    // off so ONE connection's async request burst is handled by many
    // threads (the poller-inline design serialized each conn's requests
    // behind ~ms handlers under 64-client saturation). The poller bumps
    // Conn::remain (and holds a ref) before enqueueing so an OP_SYNC
    // arriving later on the ring can never observe the writes as absent;
    // the worker balances it with finish_task after the handler returns.
    struct FastWork {
        Conn* c = nullptr;
        char op = 0;
        uint64_t seq = 0;
        std::vector<uint8_t> body;
    };
    std::mutex fast_mu_;
    std::condition_variable fast_cv_;
    std::deque<FastWork> fast_q_;
    std::vector<std::thread> fast_workers_;
    void fast_worker_main();
    void run_fast(FastWork& w);  // parse + dispatch one fast op
    void op_exchange(Conn* c, const std::vector<uint8_t>& body);
    void op_allocate(Conn* c, const RemoteMetaMsg& msg);
    void op_tcp_put(Conn* c, std::vector<uint8_t> body);
    void op_tcp_get(Conn* c, const RemoteMetaMsg& msg);
    void op_commit(Conn* c, const RemoteMetaMsg& msg);
    void op_check_exist(Conn* c, const std::vector<uint8_t>& body, const ReqCtx& ctx);
    void op_match_index(Conn* c, const std::vector<uint8_t>& body, const ReqCtx& ctx);
    void op_delete(Conn* c, const std::vector<uint8_t>& body, const ReqCtx& ctx);

    Shard* shard_for_device(int device);
    Shard* shard_least_used();
    void maybe_extend(Shard* s);
    // Evict >= `bytes` of LRU committed idle entries on `shard`. Takes
    // stripe locks itself (callers must hold none). Returns bytes freed.
    size_t evict_lru(Shard* shard, size_t bytes);
    void erase_entries(const std::vector<Ref<BlockEntry>>& entries);
    uint64_t tick() { return access_tick_.fetch_add(1, std::memory_order_relaxed); }
    static uint32_t now_sec();
    // TTL check (lazy expiry): true when the entry is past its lifetime —
    // lookups treat it as absent; the evictor reclaims it first.
    bool expired(const BlockEntry* e) const {
        return opt_.ttl_seconds > 0 &&
               now_sec() - e->born_sec > static_cast<uint32_t>(opt_.ttl_seconds);
    }
    std::atomic<uint64_t> access_tick_{1};
    std::atomic<uint64_t> n_evicted_{0};

    ServerOptions opt_;
    std::vector<std::unique_ptr<Shard>> shards_;
    std::unique_ptr<vf::Driver> vdrv_;  // lazy; created at first verbs handshake
    // Serializes driver init + arena MR registration against pool extension
    // (the extend thread registers new arenas when the driver is live).
    std::mutex vdrv_mu_;

    IoLoop main_io_;
    std::vector<std::unique_ptr<IoLoop>> workers_;
    std::atomic<uint32_t> next_worker_{0};
    uv_tcp_t listener_;
    uv_pipe_t pipe_listener_;
    bool pipe_listening_ = false;
    std::string pipe_path_;
    std::atomic<bool> running_{false};
    std::atomic<bool> stop_requested_{false};
    // TTL sweeper (only when ttl_seconds > 0): periodically erases expired
    // entries so memory frees proactively, not just on lookup/eviction.
    std::thread ttl_thread_;
    std::mutex ttl_mu_;
    std::condition_variable ttl_cv_;

    // Open-addressing key index with arena-stored keys (csrc/server/kvmap.h)
    // — node-based maps measured ~350 µs of insert cost per 2048-key write.
    // Striped key index: kStripes independent open-addressing maps, each
    // with its own reader-writer lock, selected by the top bits of the wide
    // key hash (KvMap uses the low bits for the slot). Lookups take a
    // stripe's lock shared; inserts/erases exclusive. Concurrent writers
    // land on different stripes and insert in parallel — a single exclusive
    // lock measured 334 µs avg / 8.5 ms max handler time under 8-client
    // write churn. No path ever holds two stripe locks except compact()
    // and purge(), which take them in index order.
    // 64 stripes: at 64 concurrent writers the insert pass takes exclusive
    // stripe locks; 16 stripes measured multi-ms convoys under saturation.
    static constexpr size_t kStripes = 64;
    struct KvStripe {
        // shared_mutex: dedup/read/query phases take it shared so the
        // pipelined few-connection path keeps its read concurrency. The
        // earlier "delete starvation" was NOT rwlock reader preference (a
        // plain-mutex experiment changed nothing) — it was per-block
        // allocator-lock frees under the stripe lock, fixed by extract +
        // bulk free; long sweeps now hold each stripe only briefly.
        std::shared_mutex mu;
        KvMap map;
        size_t evict_hand = 0;  // per-stripe clock cursor (mu held)
    };
    std::array<KvStripe, kStripes> kv_;
    static size_t stripe_of(uint64_t h) { return (h >> 58) & (kStripes - 1); }
    std::atomic<size_t> evict_stripe_rr_{0};

    // stats
    std::atomic<uint64_t> n_writes_{0}, n_reads_{0}, n_put_{0}, n_get_{0};
    std::atomic<uint64_t> bytes_in_{0}, bytes_out_{0};
    std::atomic<int> extending_{0};
    std::thread extend_thread_;  // managed pool extender (extend_mu_)
    std::mutex extend_mu_;

    // per-op handler timing (loop-thread time, µs) — the per-op latency log
    // the reference keeps via INFO prints (infinistore.cpp:1162-1166),
    // aggregated instead of logged per request.
    struct OpStat {
        std::atomic<uint64_t> count{0};
        std::atomic<uint64_t> total_us{0};
        std::atomic<uint64_t> max_us{0};
    };
    OpStat op_stats_[128];
};


// Connection state (one per accepted TCP client).
struct Server::Conn : RefCounted {
    Server* srv = nullptr;
    Server::IoLoop* owner = nullptr;  // the loop thread serving this conn
    // Transport handle: TCP socket or Unix-domain pipe (same-host clients
    // connect via the UDS listener — lower latency than TCP loopback).
    union {
        uv_tcp_t tcp;
        uv_pipe_t pipe;
    };
    bool is_pipe = false;
    uv_stream_t* stream() { return reinterpret_cast<uv_stream_t*>(&tcp); }
    uv_handle_t* handle() { return reinterpret_cast<uv_handle_t*>(&tcp); }
    Conn() : tcp{} {}
    bool closed = false;

    // read state machine
    enum State { kHeader, kBody } state = kHeader;
    Header hdr{};
    std::vector<uint8_t> buf;  // accumulated bytes

    // local path: in-flight async copy count. OP_SYNC blocks server-side:
    // when remain>0 the response is deferred until the count drains to zero
    // (the reference instead has the client poll with sleeps,
    // lib.py:578-592 — ~0.5 ms of added latency per op there).
    std::atomic<int> remain{0};
    bool sync_waiting = false;  // loop thread only

    // cached IPC mappings: handle bytes -> base pointer (closed on disconnect).
    // Guarded by ipc_mu: local ops for one conn can arrive concurrently on the
    // socket (owner uv-loop thread) and the shm ring (poller thread) — the
    // stock client serializes them, but server memory safety must not depend
    // on client behavior.
    std::mutex ipc_mu;
    std::map<std::vector<uint8_t>, std::pair<void*, int>> ipc_cache;  // base, src_dev
    // Handlers holding a resolved base that is not yet protected by
    // remain/fabric_inflight keep this >0; the cache flush in
    // resolve_client_base only runs when the flusher is the sole pinner.
    std::atomic<int> ipc_pin{0};

    // fabric: blocks allocated for this conn, not yet committed.
    std::unordered_map<uint64_t, Ref<BlockEntry>> pending_rdma;
    // async fabric puts in flight (owner loop defers commits behind them)
    std::atomic<int> fabric_inflight{0};
    std::vector<std::vector<uint64_t>> deferred_commits;  // owner loop only

    // verbs fabric peer (owned; torn down on the loop thread).
    Server::VerbsPeer* verbs = nullptr;

    // shared-memory ring peer (poller joined at teardown; freed in ~Conn).
    Server::ShmPeer* shm = nullptr;
    // Guards the {remain==0, sync_waiting, sync_ctx} decision — with the shm
    // transport, op_sync (poller thread) races finish_task (completion
    // threads); on the socket path both ran on the owner loop.
    std::mutex sync_mu;
    Server::ReqCtx sync_ctx;

    ~Conn() override {
        for (auto& kv : ipc_cache) {
            if (gpu::available()) gpu::ipc_close(kv.second.first);
        }
        if (shm) {
            shm->seg.unmap();
            delete shm;
        }
    }
};

}  // namespace ifs
