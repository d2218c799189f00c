#include "server.h"

#include "../core/crash.h"

#include <algorithm>
#include <chrono>
#include <cstring>
#include <future>
#include <unistd.h>

#include "../core/log.h"

namespace ifs {

// ---------------------------------------------------------------------------
// BlockEntry slab allocator (entries churn by the thousands per request)
// ---------------------------------------------------------------------------
namespace {
struct EntrySlab {
    std::mutex mu;
    std::vector<void*> free_list;
    std::vector<std::unique_ptr<char[]>> chunks;
    static constexpr size_t kPerChunk = 4096;

    void* take_locked(size_t n) {
        if (free_list.empty()) {
            chunks.emplace_back(new char[n * kPerChunk]);
            char* base = chunks.back().get();
            free_list.reserve(kPerChunk);
            for (size_t i = 0; i < kPerChunk; i++) free_list.push_back(base + i * n);
        }
        void* p = free_list.back();
        free_list.pop_back();
        return p;
    }

    // Thread-local magazines: entry alloc/free happens on every handler,
    // completion and delete path — one global mutex per op (especially
    // give(), which delete churn hit ~6M times per 64-client run) serialized
    // the whole server. Each thread caches up to kMagMax entries and
    // exchanges them with the global list kMagBatch at a time.
    static constexpr size_t kMagBatch = 256;
    static constexpr size_t kMagMax = 512;
    struct Magazine {
        std::vector<void*> items;
        ~Magazine();  // flush to the global list (pollers exit per conn)
    };
    static Magazine& mag() {
        thread_local Magazine m;
        return m;
    }

    void* take(size_t n) {
        auto& m = mag();
        if (m.items.empty()) {
            std::lock_guard<std::mutex> lk(mu);
            m.items.reserve(kMagMax);
            for (size_t i = 0; i < kMagBatch; i++) m.items.push_back(take_locked(n));
        }
        void* p = m.items.back();
        m.items.pop_back();
        return p;
    }
    // Bulk take for a write batch: drain the magazine, top up from the
    // global list under ONE lock.
    void take_bulk(size_t n, size_t count, std::vector<void*>* out) {
        auto& m = mag();
        while (count && !m.items.empty()) {
            out->push_back(m.items.back());
            m.items.pop_back();
            count--;
        }
        if (count) {
            std::lock_guard<std::mutex> lk(mu);
            for (size_t i = 0; i < count; i++) out->push_back(take_locked(n));
        }
    }
    void give(void* p) {
        auto& m = mag();
        m.items.push_back(p);
        if (m.items.size() >= kMagMax) {
            std::lock_guard<std::mutex> lk(mu);
            for (size_t i = 0; i < kMagBatch; i++) {
                free_list.push_back(m.items.back());
                m.items.pop_back();
            }
        }
    }
    void give_bulk(std::vector<void*>& items, size_t from) {
        auto& m = mag();
        for (size_t i = from; i < items.size(); i++) m.items.push_back(items[i]);
        if (m.items.size() >= kMagMax) {
            std::lock_guard<std::mutex> lk(mu);
            while (m.items.size() > kMagBatch) {
                free_list.push_back(m.items.back());
                m.items.pop_back();
            }
        }
    }
};
EntrySlab& entry_slab() {
    // Intentionally leaked: entries may still be freed during interpreter
    // teardown after static destructors would have run.
    static EntrySlab* s = new EntrySlab();
    return *s;
}

EntrySlab::Magazine::~Magazine() {
    if (items.empty()) return;
    auto& s = entry_slab();  // leaked singleton: safe at thread exit
    std::lock_guard<std::mutex> lk(s.mu);
    for (void* p : items) s.free_list.push_back(p);
}
}  // namespace

void* BlockEntry::operator new(size_t n) { return entry_slab().take(n); }
void BlockEntry::operator delete(void* p) { entry_slab().give(p); }

namespace {
// One lock for a whole burst of entry allocations (hot write path).
struct SlabBatch {
    std::vector<void*> slots;
    size_t next = 0;
    explicit SlabBatch(size_t count) {
        slots.reserve(count);
        entry_slab().take_bulk(sizeof(BlockEntry), count, &slots);
    }
    ~SlabBatch() {
        if (next >= slots.size()) return;
        entry_slab().give_bulk(slots, next);
    }
    BlockEntry* make() { return new (slots[next++]) BlockEntry(); }
};
}  // namespace

namespace {

struct WriteReq {
    uv_write_t req;
    std::vector<uint8_t> data;
    std::shared_ptr<uint8_t[]> raw;  // alternative payload (uninitialized buf)
    size_t raw_len = 0;
};

void send_buf(Server::Conn* c, std::vector<uint8_t> data);

void conn_close(Server::Conn* c) {
    if (c->closed) return;
    c->closed = true;
    uv_close(c->handle(), [](uv_handle_t* h) {
        auto* c = static_cast<Server::Conn*>(h->data);
        c->unref();
    });
}

void send_status(Server::Conn* c, int code) {
    std::vector<uint8_t> v(4);
    memcpy(v.data(), &code, 4);
    send_buf(c, std::move(v));
}

void send_status_payload(Server::Conn* c, int code, const uint8_t* payload, size_t n) {
    std::vector<uint8_t> v(8 + n);
    uint32_t len = static_cast<uint32_t>(n);
    memcpy(v.data(), &code, 4);
    memcpy(v.data() + 4, &len, 4);
    if (n) memcpy(v.data() + 8, payload, n);
    send_buf(c, std::move(v));
}

// Reference-framed response: status + raw payload with NO length prefix
// (the reference's send_resp, infinistore.cpp:1055-1068 — its clients know
// each op's payload size a priori). Used by the query ops for byte-level
// wire compatibility.
void send_status_raw(Server::Conn* c, int code, const void* payload, size_t n) {
    std::vector<uint8_t> v(4 + n);
    memcpy(v.data(), &code, 4);
    if (n) memcpy(v.data() + 4, payload, n);
    send_buf(c, std::move(v));
}

void send_raw(Server::Conn* c, std::shared_ptr<uint8_t[]> buf, size_t len) {
    if (c->closed) return;
    auto* wr = new WriteReq();
    wr->raw = std::move(buf);
    wr->raw_len = len;
    wr->req.data = wr;
    uv_buf_t b = uv_buf_init(reinterpret_cast<char*>(wr->raw.get()),
                             static_cast<unsigned>(len));
    int r = uv_write(&wr->req, c->stream(), &b, 1,
                     [](uv_write_t* req, int status) {
                         auto* wr2 = static_cast<WriteReq*>(req->data);
                         delete wr2;
                         if (status < 0) DEBUG("uv_write status %d", status);
                     });
    if (r != 0) {
        delete wr;
        conn_close(c);
    }
}

void send_buf(Server::Conn* c, std::vector<uint8_t> data) {
    if (c->closed) return;
    auto* wr = new WriteReq();
    wr->data = std::move(data);
    wr->req.data = wr;
    uv_buf_t b = uv_buf_init(reinterpret_cast<char*>(wr->data.data()),
                             static_cast<unsigned>(wr->data.size()));
    int r = uv_write(&wr->req, c->stream(), &b, 1,
                     [](uv_write_t* req, int status) {
                         auto* wr = static_cast<WriteReq*>(req->data);
                         delete wr;
                         if (status < 0) DEBUG("uv_write status %d", status);
                     });
    if (r != 0) {
        delete wr;
        conn_close(c);
    }
}

}  // namespace

// ---------------------------------------------------------------------------
// Server lifecycle
// ---------------------------------------------------------------------------
Server::Server(const ServerOptions& opt) : opt_(opt) {
    set_log_level(opt.log_level.c_str());
    if (opt_.devices.empty()) {
        for (int i = 0; i < std::max(1, opt_.cpu_shards); i++) {
            ShardOptions so;
            so.device = -1;
            so.pool_bytes = opt_.prealloc_bytes;
            so.block_granule = opt_.block_granule;
            so.auto_extend = opt_.auto_extend;
            so.extend_bytes = opt_.extend_bytes;
            shards_.emplace_back(new Shard(so));
        }
    } else {
        for (int dev : opt_.devices) {
            ShardOptions so;
            so.device = dev;
            so.pool_bytes = opt_.prealloc_bytes;
            so.block_granule = opt_.block_granule;
            so.n_streams = opt_.n_streams;
            so.auto_extend = opt_.auto_extend;
            so.extend_bytes = opt_.extend_bytes;
            shards_.emplace_back(new Shard(so));
        }
    }
}

Server::~Server() { stop(); }

bool Server::start() {
    if (running_.load()) return true;
    install_crash_handlers();
    for (auto& s : shards_) {
        if (!s->init()) {
            ERROR("shard init failed");
            return false;
        }
    }
    // Enable xGMI peer access between all shard devices up front (cross-GPU
    // reads/writes run the copy kernel on the pool device against peer VAs).
    if (gpu::available()) {
        int n = gpu::device_count();
        for (auto& s : shards_) {
            if (!s->on_gpu()) continue;
            for (int peer = 0; peer < n; peer++) {
                if (peer != s->device()) gpu::enable_peer_access(s->device(), peer);
            }
        }
    }

    // Bring up the main IO loop (listener) and the worker IO loops.
    main_io_.srv = this;
    main_io_.is_main = true;
    uv_loop_init(&main_io_.loop);
    uv_async_init(&main_io_.loop, &main_io_.post_async, &IoLoop::on_post);
    main_io_.post_async.data = &main_io_;
    uv_async_init(&main_io_.loop, &main_io_.stop_async, &IoLoop::on_stop);
    main_io_.stop_async.data = &main_io_;

    uv_tcp_init(&main_io_.loop, &listener_);
    listener_.data = this;
    struct sockaddr_in addr;
    uv_ip4_addr("0.0.0.0", opt_.service_port, &addr);
    int r = uv_tcp_bind(&listener_, reinterpret_cast<const struct sockaddr*>(&addr), 0);
    if (r == 0)
        r = uv_listen(reinterpret_cast<uv_stream_t*>(&listener_), 512,
                      &Server::on_new_connection);
    if (r != 0) {
        ERROR("bind/listen 0.0.0.0:%d failed: %s", opt_.service_port, uv_strerror(r));
        uv_close(reinterpret_cast<uv_handle_t*>(&listener_), nullptr);
        uv_close(reinterpret_cast<uv_handle_t*>(&main_io_.post_async), nullptr);
        uv_close(reinterpret_cast<uv_handle_t*>(&main_io_.stop_async), nullptr);
        uv_run(&main_io_.loop, UV_RUN_NOWAIT);
        uv_loop_close(&main_io_.loop);
        return false;
    }
    // Same-host fast transport: a Unix-domain listener next to the TCP one.
    pipe_path_ = "/tmp/infinistore-amd-" + std::to_string(opt_.service_port) + ".sock";
    unlink(pipe_path_.c_str());
    uv_pipe_init(&main_io_.loop, &pipe_listener_, 0);
    pipe_listener_.data = this;
    if (uv_pipe_bind(&pipe_listener_, pipe_path_.c_str()) == 0 &&
        uv_listen(reinterpret_cast<uv_stream_t*>(&pipe_listener_), 512,
                  &Server::on_new_pipe_connection) == 0) {
        pipe_listening_ = true;
    } else {
        WARN("UDS listener at %s unavailable; same-host clients use TCP",
             pipe_path_.c_str());
        uv_close(reinterpret_cast<uv_handle_t*>(&pipe_listener_), nullptr);
    }

    for (int i = 0; i < std::max(0, opt_.io_threads); i++) {
        auto w = std::make_unique<IoLoop>();
        w->srv = this;
        uv_loop_init(&w->loop);
        uv_async_init(&w->loop, &w->post_async, &IoLoop::on_post);
        w->post_async.data = w.get();
        uv_async_init(&w->loop, &w->stop_async, &IoLoop::on_stop);
        w->stop_async.data = w.get();
        workers_.push_back(std::move(w));
    }
    // Pre-size the index for the pool's block capacity: every block can
    // carry a key, and running the maps near their load limit means rehash
    // + tombstone-compaction cycles under the exclusive stripe locks (a
    // 64-client churn run measured 11 µs per lookup from probe-chain
    // growth). Capped at 16M slots (~1 GB of index).
    // IFS_KV_INITIAL (tests only) shrinks the initial capacity so rehash and
    // tombstone-compaction paths are exercised by small workloads.
    size_t pool_blocks = 0;
    for (auto& s : shards_) pool_blocks += s->total_blocks();
    size_t kv_initial = std::max<size_t>(1u << 20, pool_blocks + pool_blocks / 2);
    kv_initial = std::min<size_t>(kv_initial, 1u << 24);
    if (const char* env = getenv("IFS_KV_INITIAL")) kv_initial = strtoull(env, nullptr, 10);
    for (auto& st : kv_) st.map.reserve(kv_initial / kStripes);
    running_.store(true);
    stop_requested_.store(false);
    // Fast-op workers (ring write/read handlers): enough to keep every HIP
    // stream fed under a 64-client burst without oversubscribing the box.
    {
        unsigned hw = std::thread::hardware_concurrency();
        unsigned n = std::min(24u, std::max(6u, hw / 8));
        if (const char* e = getenv("IFS_FAST_WORKERS")) {
            unsigned v = static_cast<unsigned>(strtoul(e, nullptr, 10));
            if (v >= 1 && v <= 128) n = v;
        }
        for (unsigned i = 0; i < n; i++)
            fast_workers_.emplace_back([this] { fast_worker_main(); });
    }
    main_io_.start();
    for (auto& w : workers_) w->start();
    if (opt_.ttl_seconds > 0) {
        ttl_thread_ = std::thread([this] {
            // Sweep every ttl/4 (>=1 s): one bounded pass over each stripe,
            // erasing expired idle entries. Lookups already treat expired
            // keys as absent; this just returns their memory early.
            auto period = std::chrono::seconds(std::max(1, opt_.ttl_seconds / 4));
            std::unique_lock<std::mutex> lk(ttl_mu_);
            while (!ttl_cv_.wait_for(lk, period,
                                     [this] { return stop_requested_.load(); })) {
                lk.unlock();
                size_t swept = 0;
                for (auto& st : kv_) {
                    std::vector<Ref<BlockEntry>> dead;  // dropped after unlock
                    {
                        std::lock_guard<std::shared_mutex> ex(st.mu);
                        std::vector<std::string> victims;
                        st.map.for_each([&](std::string_view key, Ref<BlockEntry>& val) {
                            if (expired(val.get()) && val->ref_count() == 1)
                                victims.emplace_back(key);
                        });
                        dead.reserve(victims.size());
                        for (auto& k : victims) {
                            Ref<BlockEntry> ref;
                            if (st.map.extract(k, &ref)) dead.emplace_back(std::move(ref));
                        }
                        swept += victims.size();
                    }
                }
                if (swept) DEBUG("ttl sweep: %zu expired entries erased", swept);
                lk.lock();
            }
        });
    }
    INFO("server listening on 0.0.0.0:%d with %zu shard(s), %zu IO loops",
         opt_.service_port, shards_.size(), workers_.size() + 1);
    return true;
}

void Server::IoLoop::start() {
    thread = std::thread([this] {
        uv_run(&loop, UV_RUN_DEFAULT);
        uv_loop_close(&loop);
    });
}

void Server::stop() {
    if (!running_.load()) return;
    stop_requested_.store(true);
    ttl_cv_.notify_all();
    if (ttl_thread_.joinable()) ttl_thread_.join();
    for (auto& w : workers_) w->request_stop();
    main_io_.request_stop();
    for (auto& w : workers_)
        if (w->thread.joinable()) w->thread.join();
    if (main_io_.thread.joinable()) main_io_.thread.join();
    workers_.clear();
    // After the IO loops (and with them every ring poller) are joined no
    // new fast work can arrive; drain the queue, then join the workers.
    fast_cv_.notify_all();
    for (auto& t : fast_workers_)
        if (t.joinable()) t.join();
    fast_workers_.clear();
    {
        std::lock_guard<std::mutex> jl(extend_mu_);
        if (extend_thread_.joinable()) extend_thread_.join();
    }
    running_.store(false);
    // Drop all stored blocks.
    purge();
}

void Server::IoLoop::on_stop(uv_async_t* h) {
    auto* io = static_cast<IoLoop*>(h->data);
    // Close this loop's connections and handles; the loop exits when none
    // remain.
    for (auto* c : io->conns) {
        io->srv->verbs_teardown(c);
        io->srv->shm_teardown(c);
        conn_close(c);
    }
    io->conns.clear();
    if (io->is_main) {
        uv_close(reinterpret_cast<uv_handle_t*>(&io->srv->listener_), nullptr);
        if (io->srv->pipe_listening_) {
            uv_close(reinterpret_cast<uv_handle_t*>(&io->srv->pipe_listener_), nullptr);
            unlink(io->srv->pipe_path_.c_str());
            io->srv->pipe_listening_ = false;
        }
    }
    uv_close(reinterpret_cast<uv_handle_t*>(&io->post_async), nullptr);
    uv_close(reinterpret_cast<uv_handle_t*>(&io->stop_async), nullptr);
}

void Server::IoLoop::post(std::function<void()> fn) {
    // After stop is requested the async handles are being closed; drop the
    // callback (captured refs release on destruction) instead of touching a
    // dying handle.
    if (srv && srv->stop_requested_.load(std::memory_order_acquire)) return;
    {
        std::lock_guard<std::mutex> lk(post_mu);
        posted.push_back(std::move(fn));
    }
    uv_async_send(&post_async);
}

void Server::IoLoop::on_post(uv_async_t* h) {
    auto* io = static_cast<IoLoop*>(h->data);
    std::vector<std::function<void()>> fns;
    {
        std::lock_guard<std::mutex> lk(io->post_mu);
        fns.swap(io->posted);
    }
    for (auto& f : fns) f();
}

// ---------------------------------------------------------------------------
// Accept + read state machine
// ---------------------------------------------------------------------------
void Server::on_new_connection(uv_stream_t* server, int status) {
    auto* srv = static_cast<Server*>(server->data);
    if (status < 0) {
        WARN("accept error: %s", uv_strerror(status));
        return;
    }
    // Accept on the main loop, then hand the fd to a worker loop (round
    // robin) so request handling scales across IO threads.
    uv_tcp_t* tmp = new uv_tcp_t();
    uv_tcp_init(&srv->main_io_.loop, tmp);
    tmp->data = nullptr;
    if (uv_accept(server, reinterpret_cast<uv_stream_t*>(tmp)) != 0) {
        uv_close(reinterpret_cast<uv_handle_t*>(tmp),
                 [](uv_handle_t* h) { delete reinterpret_cast<uv_tcp_t*>(h); });
        return;
    }
    uv_os_fd_t fd;
    if (uv_fileno(reinterpret_cast<uv_handle_t*>(tmp), &fd) != 0) {
        uv_close(reinterpret_cast<uv_handle_t*>(tmp),
                 [](uv_handle_t* h) { delete reinterpret_cast<uv_tcp_t*>(h); });
        return;
    }
    int fd2 = dup(fd);
    uv_close(reinterpret_cast<uv_handle_t*>(tmp),
             [](uv_handle_t* h) { delete reinterpret_cast<uv_tcp_t*>(h); });
    IoLoop* io = &srv->main_io_;
    if (!srv->workers_.empty()) {
        uint32_t i = srv->next_worker_.fetch_add(1) %
                     static_cast<uint32_t>(srv->workers_.size());
        io = srv->workers_[i].get();
    }
    if (io == &srv->main_io_)
        srv->adopt_fd(io, fd2, false);
    else
        io->post([srv, io, fd2] { srv->adopt_fd(io, fd2, false); });
}

void Server::on_new_pipe_connection(uv_stream_t* server, int status) {
    auto* srv = static_cast<Server*>(server->data);
    if (status < 0) return;
    uv_pipe_t* tmp = new uv_pipe_t();
    uv_pipe_init(&srv->main_io_.loop, tmp, 0);
    tmp->data = nullptr;
    auto close_tmp = [](uv_handle_t* h) { delete reinterpret_cast<uv_pipe_t*>(h); };
    if (uv_accept(server, reinterpret_cast<uv_stream_t*>(tmp)) != 0) {
        uv_close(reinterpret_cast<uv_handle_t*>(tmp), close_tmp);
        return;
    }
    uv_os_fd_t fd;
    if (uv_fileno(reinterpret_cast<uv_handle_t*>(tmp), &fd) != 0) {
        uv_close(reinterpret_cast<uv_handle_t*>(tmp), close_tmp);
        return;
    }
    int fd2 = dup(fd);
    uv_close(reinterpret_cast<uv_handle_t*>(tmp), close_tmp);
    IoLoop* io = &srv->main_io_;
    if (!srv->workers_.empty()) {
        uint32_t i = srv->next_worker_.fetch_add(1) %
                     static_cast<uint32_t>(srv->workers_.size());
        io = srv->workers_[i].get();
    }
    if (io == &srv->main_io_)
        srv->adopt_fd(io, fd2, true);
    else
        io->post([srv, io, fd2] { srv->adopt_fd(io, fd2, true); });
}

void Server::adopt_fd(IoLoop* io, int fd, bool is_pipe) {
    if (stop_requested_.load()) {
        ::close(fd);
        return;
    }
    int bufsz = 4 << 20;  // bulk fabric payloads ride this socket
    setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &bufsz, sizeof(bufsz));
    setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &bufsz, sizeof(bufsz));
    auto* c = new Conn();
    c->srv = this;
    c->owner = io;
    c->is_pipe = is_pipe;
    if (is_pipe) {
        uv_pipe_init(&io->loop, &c->pipe, 0);
        c->pipe.data = c;
        if (uv_pipe_open(&c->pipe, fd) != 0) {
            conn_close(c);
            return;
        }
    } else {
        uv_tcp_init(&io->loop, &c->tcp);
        c->tcp.data = c;
        if (uv_tcp_open(&c->tcp, fd) != 0) {
            conn_close(c);
            return;
        }
        uv_tcp_nodelay(&c->tcp, 1);
    }
    io->conns.push_back(c);
    uv_read_start(
        c->stream(),
        [](uv_handle_t*, size_t suggested, uv_buf_t* buf) {
            buf->base = static_cast<char*>(malloc(suggested));
            buf->len = suggested;
        },
        [](uv_stream_t* stream, ssize_t nread, const uv_buf_t* buf) {
            auto* c = static_cast<Conn*>(stream->data);
            Server* srv = c->srv;
            if (nread < 0) {
                free(buf->base);
                auto& v = c->owner->conns;
                v.erase(std::remove(v.begin(), v.end(), c), v.end());
                srv->verbs_teardown(c);
                srv->shm_teardown(c);
                conn_close(c);
                return;
            }
            if (nread == 0) {
                free(buf->base);
                return;
            }
            c->buf.insert(c->buf.end(), reinterpret_cast<uint8_t*>(buf->base),
                          reinterpret_cast<uint8_t*>(buf->base) + nread);
            free(buf->base);
            // Drain as many complete requests as are buffered.
            size_t consumed = 0;
            for (;;) {
                if (c->state == Conn::kHeader) {
                    if (c->buf.size() - consumed < sizeof(Header)) break;
                    memcpy(&c->hdr, c->buf.data() + consumed, sizeof(Header));
                    consumed += sizeof(Header);
                    if (c->hdr.magic != kMagic) {
                        WARN("bad magic from client; closing");
                        auto& v = c->owner->conns;
                        v.erase(std::remove(v.begin(), v.end(), c), v.end());
                        srv->verbs_teardown(c);
                        srv->shm_teardown(c);
                        conn_close(c);
                        return;
                    }
                    size_t cap = (c->hdr.op == OP_TCP_PUT) ? (256u << 20) : kProtocolBufferSize;
                    if (c->hdr.body_size > cap) {
                        WARN("body too large (%u) for op %c", c->hdr.body_size, c->hdr.op);
                        auto& v = c->owner->conns;
                        v.erase(std::remove(v.begin(), v.end(), c), v.end());
                        srv->verbs_teardown(c);
                        srv->shm_teardown(c);
                        conn_close(c);
                        return;
                    }
                    c->state = Conn::kBody;
                }
                if (c->state == Conn::kBody) {
                    if (c->buf.size() - consumed < c->hdr.body_size) break;
                    std::vector<uint8_t> body(c->buf.data() + consumed,
                                              c->buf.data() + consumed + c->hdr.body_size);
                    consumed += c->hdr.body_size;
                    c->state = Conn::kHeader;
                    srv->handle_request(c, c->hdr.op, std::move(body));
                    if (c->closed) return;
                }
            }
            if (consumed) c->buf.erase(c->buf.begin(), c->buf.begin() + consumed);
        });
}

// ---------------------------------------------------------------------------
// Dispatch
// ---------------------------------------------------------------------------
namespace {

// View over the flatbuffers LocalMetaRequest (msg must outlive the view).
Server::LocalView to_view(const LocalMetaMsg& msg) {
    Server::LocalView v;
    v.device = msg.device;
    v.pid = msg.pid;
    v.base_ptr = msg.base_ptr;
    v.base_offset = msg.base_offset;
    v.block_size = msg.block_size;
    v.ipc = msg.ipc_handle.data();
    v.ipc_len = msg.ipc_handle.size();
    v.blocks.reserve(msg.blocks.size());
    for (auto& b : msg.blocks) v.blocks.push_back({std::string_view(b.key), b.offset});
    return v;
}

// Packed fast-path body: PackedLocalHdr, u64 offsets[n], NUL-separated keys.
bool parse_packed_local(const uint8_t* body, size_t body_len, Server::LocalView* v) {
    if (body_len < sizeof(PackedLocalHdr)) return false;
    PackedLocalHdr h;
    memcpy(&h, body, sizeof(h));
    size_t n = h.n_blocks;
    size_t off_end = sizeof(h) + n * 8;
    if (body_len < off_end) return false;
    v->device = h.device;
    v->pid = h.pid;
    v->base_ptr = h.base_ptr;
    v->base_offset = h.base_offset;
    v->block_size = h.block_size;
    v->flags = h.flags;
    v->alloc_mb = h.rsvd;
    v->ipc = body + offsetof(PackedLocalHdr, ipc);
    v->ipc_len = 64;
    const uint64_t* offs = reinterpret_cast<const uint64_t*>(body + sizeof(h));
    const char* kp = reinterpret_cast<const char*>(body + off_end);
    const char* kend = reinterpret_cast<const char*>(body + body_len);
    v->blocks.reserve(n);
    for (size_t i = 0; i < n; i++) {
        const char* nul = static_cast<const char*>(memchr(kp, 0, kend - kp));
        const char* ke = nul ? nul : kend;
        if (kp > kend || (i + 1 < n && !nul)) return false;
        v->blocks.push_back({std::string_view(kp, ke - kp), offs[i]});
        kp = nul ? nul + 1 : kend;
    }
    return true;
}

}  // namespace

void Server::handle_request(Conn* c, char op, std::vector<uint8_t> body) {
    DEBUG("request op=%s body=%zu", op_name(op).c_str(), body.size());
    struct Timer {
        Server* s;
        char op;
        std::chrono::steady_clock::time_point t0 = std::chrono::steady_clock::now();
        ~Timer() {
            auto us = std::chrono::duration_cast<std::chrono::microseconds>(
                          std::chrono::steady_clock::now() - t0)
                          .count();
            auto& st = s->op_stats_[static_cast<uint8_t>(op) & 127];
            st.count.fetch_add(1, std::memory_order_relaxed);
            st.total_us.fetch_add(static_cast<uint64_t>(us), std::memory_order_relaxed);
            uint64_t prev = st.max_us.load(std::memory_order_relaxed);
            while (static_cast<uint64_t>(us) > prev &&
                   !st.max_us.compare_exchange_weak(prev, static_cast<uint64_t>(us))) {
            }
        }
    } timer{this, op};
    switch (op) {
        case OP_W: {
            LocalMetaMsg msg;
            if (!parse_local_meta(body.data(), body.size(), &msg)) return send_status(c, INVALID_REQ);
            return op_local_write(c, to_view(msg), ReqCtx{});
        }
        case OP_R: {
            LocalMetaMsg msg;
            if (!parse_local_meta(body.data(), body.size(), &msg)) return send_status(c, INVALID_REQ);
            return op_local_read(c, to_view(msg), ReqCtx{});
        }
        case OP_W_FAST: {
            LocalView v;
            if (!parse_packed_local(body.data(), body.size(), &v))
                return send_status(c, INVALID_REQ);
            return op_local_write(c, v, ReqCtx{});
        }
        case OP_R_FAST: {
            LocalView v;
            if (!parse_packed_local(body.data(), body.size(), &v))
                return send_status(c, INVALID_REQ);
            return op_local_read(c, v, ReqCtx{});
        }
        case OP_SYNC:
            return op_sync(c, ReqCtx{});
        case OP_SHM_SETUP:
            return op_shm_setup(c, body);
        case OP_RDMA_EXCHANGE:
            return op_exchange(c, body);
        case OP_RDMA_ALLOCATE: {
            RemoteMetaMsg msg;
            if (!parse_remote_meta(body.data(), body.size(), &msg))
                return send_status(c, INVALID_REQ);
            return op_allocate(c, msg);
        }
        case OP_TCP_PUT:
            return op_tcp_put(c, std::move(body));
        case OP_TCP_GET: {
            RemoteMetaMsg msg;
            if (!parse_remote_meta(body.data(), body.size(), &msg))
                return send_status(c, INVALID_REQ);
            return op_tcp_get(c, msg);
        }
        case OP_RDMA_WRITE_COMMIT: {
            RemoteMetaMsg msg;
            if (!parse_remote_meta(body.data(), body.size(), &msg))
                return send_status(c, INVALID_REQ);
            return op_commit(c, msg);
        }
        case OP_CHECK_EXIST:
            return op_check_exist(c, body, ReqCtx{});
        case OP_GET_MATCH_LAST_IDX:
            return op_match_index(c, body, ReqCtx{});
        case OP_DELETE:
            return op_delete(c, body, ReqCtx{});
        case OP_STATS: {
            std::string js = stats_json();
            return send_status_payload(c, FINISH,
                                       reinterpret_cast<const uint8_t*>(js.data()), js.size());
        }
        default:
            WARN("unknown op '%c'", op);
            return send_status(c, INVALID_REQ);
    }
}

Shard* Server::shard_for_device(int device) {
    for (auto& s : shards_)
        if (s->device() == device) return s.get();
    return shards_[static_cast<size_t>(device < 0 ? 0 : device) % shards_.size()].get();
}

Shard* Server::shard_least_used() {
    Shard* best = shards_[0].get();
    size_t best_used = SIZE_MAX;
    for (auto& s : shards_) {
        size_t used = s->used_blocks();
        if (used < best_used) {
            best_used = used;
            best = s.get();
        }
    }
    return best;
}

uint32_t Server::now_sec() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return static_cast<uint32_t>(ts.tv_sec);
}

void Server::erase_entries(const std::vector<Ref<BlockEntry>>& entries) {
    // Rare path (allocation failure / failed copy): remove entries from the
    // index by identity. O(map) scan, but keys are not kept around on the
    // hot path. One stripe at a time.
    for (auto& st : kv_) {
        std::vector<Ref<BlockEntry>> dead;  // dropped after unlock
        {
            std::lock_guard<std::shared_mutex> lk(st.mu);
            std::vector<std::string> victims;
            st.map.for_each([&](std::string_view key, Ref<BlockEntry>& val) {
                for (auto& e : entries)
                    if (val.get() == e.get()) victims.emplace_back(key);
            });
            dead.reserve(victims.size());
            for (auto& k : victims) {
                Ref<BlockEntry> ref;
                if (st.map.extract(k, &ref)) dead.emplace_back(std::move(ref));
            }
        }
    }
}

size_t Server::evict_lru(Shard* shard, size_t bytes) {
    // Sampled clock-hand eviction: scan bounded slot windows from a
    // persistent per-stripe cursor, evict the least-recently-accessed half
    // of each sample — O(evicted) amortized instead of a full index scan
    // per eviction event (which capped sustained full-pool churn).
    // Candidates: committed, idle (only the map holds a ref), on this shard.
    // Key views stay valid across erase (the arena is append-only).
    // Stripes are visited round-robin (shared cursor) holding one stripe
    // lock at a time.
    size_t freed = 0;
    // NOTE on fresh writes: the sample sort below already evicts oldest
    // first, so recently written keys survive unless a sample is entirely
    // fresh (possible at ~100% occupancy — the pool is then genuinely too
    // small for the working set and evicting fresh data is the honest LRU
    // outcome). An explicit freshness-exemption pass was tried and reverted:
    // at full occupancy it degenerated to an O(map) rescan per allocation
    // (13x soak throughput collapse).
    for (size_t visited = 0; freed < bytes && visited < kStripes; visited++) {
        auto& st = kv_[evict_stripe_rr_.fetch_add(1) % kStripes];
        std::vector<Ref<BlockEntry>> dead;  // block frees run after unlock
        std::lock_guard<std::shared_mutex> lk(st.mu);
        size_t scanned = 0;
        const size_t scan_cap = st.map.capacity();  // one revolution max
        std::vector<std::pair<uint64_t, std::string_view>> sample;
        sample.reserve(128);
        while (freed < bytes && scanned < scan_cap) {
            size_t window = 4096;
            st.map.scan_from(&st.evict_hand, window,
                             [&](std::string_view key, Ref<BlockEntry>& val) {
                                 BlockEntry* e = val.get();
                                 uint64_t la = e->last_access.load(std::memory_order_relaxed);
                                 if (e->shard == shard && e->committed && e->ref_count() == 1)
                                     sample.push_back(
                                         {expired(e) ? 0  // expired: evict first
                                                     : la,
                                          key});
                                 return sample.size() < 128;
                             });
            scanned += window;
            // In a sparse table one window yields few candidates; evicting
            // from a 1-2 entry "sample" degrades LRU to "next key after the
            // cursor". Accumulate across windows until the sample is big
            // enough to rank (or the stripe is fully scanned).
            if (sample.size() < 128 && scanned < scan_cap) continue;
            if (sample.empty()) break;
            std::sort(sample.begin(), sample.end(),
                      [](const auto& a, const auto& b) { return a.first < b.first; });
            size_t take = std::max<size_t>(1, sample.size() / 2);
            for (size_t i = 0; i < take && freed < bytes; i++) {
                Ref<BlockEntry> ref;
                if (!st.map.extract(sample[i].second, &ref)) continue;
                freed += ref->size;
                dead.emplace_back(std::move(ref));
                n_evicted_.fetch_add(1);
            }
            sample.clear();
        }
    }
    if (freed) DEBUG("auto-evicted %zu bytes from shard dev=%d", freed, shard->device());
    return freed;
}

void Server::maybe_extend(Shard* s) {
    if (!s->need_extend()) return;
    int expect = 0;
    if (!extending_.compare_exchange_strong(expect, 1)) return;
    // Managed (not detached): a detached extender could still be unlocking
    // vdrv_mu_ while ~Server destroys it (TSAN-caught shutdown race).
    std::lock_guard<std::mutex> jl(extend_mu_);
    if (extend_thread_.joinable()) extend_thread_.join();  // prior run done
    extend_thread_ = std::thread([this, s] {
        INFO("extending pool on shard dev=%d", s->device());
        void* arena = nullptr;
        if (s->extend(&arena) && arena) {
            // Verbs fabric live: the new arena must carry an MR or verbs
            // reads from it fail ("pool arena not registered"); the reference
            // registers an MR per pool at pool creation (mempool.cpp:29-44).
            std::lock_guard<std::mutex> lk(vdrv_mu_);
            if (vdrv_ && !vdrv_->reg_region(arena, opt_.extend_bytes, s->on_gpu(), nullptr))
                ERROR("MR registration of extended arena failed; verbs reads from it will error");
        }
        extending_.store(0);
    });
}

// ---- local (IPC) path -----------------------------------------------------
namespace {
// Resolve the client's allocation base: same-process fast path (an IPC
// handle cannot be opened inside the exporting process) or the per-conn
// cached hipIpcOpenMemHandle mapping.
void* resolve_client_base(Server::Conn* c, const Server::LocalView& msg) {
    if (msg.pid != 0 && msg.pid == static_cast<int32_t>(getpid()) && msg.base_ptr != 0)
        return reinterpret_cast<void*>(msg.base_ptr);
    // dmabuf IPC limitation on this ROCm/driver stack: hipIpcOpenMemHandle
    // HANGS FOREVER importing allocations >= 2 GiB (bisected by
    // scripts/ipc_size_probe.py: 1.5 GiB fine, 2.0 GiB hangs). Refuse
    // instead of wedging this conn's poller/worker thread. Clients see a
    // clear error and can split tensors (< 2 GiB per allocation, e.g.
    // per-layer KV pools) or use the fabric path for the big tensor.
    if (msg.alloc_mb >= 2048) {
        ERROR("refusing IPC import of a %u MB client allocation: "
              "hipIpcOpenMemHandle hangs for >= 2 GiB allocations under "
              "dmabuf IPC; split the tensor or use the fabric path",
              msg.alloc_mb);
        return nullptr;
    }
    std::vector<uint8_t> key(msg.ipc, msg.ipc + msg.ipc_len);
    std::lock_guard<std::mutex> lk(c->ipc_mu);
    auto it = c->ipc_cache.find(key);
    if (it != c->ipc_cache.end()) return it->second.first;
    // Bound the cache: a client that churns tensors would otherwise pin
    // every old allocation via its stale mapping. Only flush when nothing
    // is in flight (a mapping may be read by a queued kernel).
    if (c->ipc_cache.size() >= 64 && c->remain.load() == 0 &&
        c->fabric_inflight.load() == 0 && c->ipc_pin.load() == 1) {
        for (auto& kv2 : c->ipc_cache) gpu::ipc_close(kv2.second.first);
        c->ipc_cache.clear();
    }
    gpu::IpcHandle h;
    memcpy(h.bytes, msg.ipc, gpu::kIpcHandleSize);
    void* base = gpu::ipc_open(h, msg.device);
    if (base) c->ipc_cache.emplace(std::move(key), std::make_pair(base, msg.device));
    return base;
}

// Holds Conn::ipc_pin for a handler's resolve→enqueue window so a concurrent
// handler on the other transport cannot flush (ipc_close) a base this one
// resolved but has not yet protected with remain/fabric_inflight.
struct IpcPinGuard {
    Server::Conn* c;
    explicit IpcPinGuard(Server::Conn* conn) : c(conn) { c->ipc_pin.fetch_add(1); }
    ~IpcPinGuard() { c->ipc_pin.fetch_sub(1); }
};
}  // namespace

void Server::op_local_write(Conn* c, const LocalView& msg, const ReqCtx& ctx) {
    static const bool pdbg = getenv("IFS_SERVER_DEBUG") != nullptr;
    auto p0 = std::chrono::steady_clock::now();
    if (!gpu::available()) return reply_local(c, ctx, SYSTEM_ERROR);
    if (msg.ipc_len != gpu::kIpcHandleSize || msg.block_size <= 0)
        return reply_local(c, ctx, INVALID_REQ);
    IpcPinGuard pin(c);
    void* base = resolve_client_base(c, msg);
    if (!base) return reply_local(c, ctx, INTERNAL_ERROR);
    uint8_t* client_ptr = static_cast<uint8_t*>(base) + msg.base_offset;

    Shard* shard = shard_for_device(msg.device);
    size_t page = static_cast<size_t>(msg.block_size);
    size_t nb = msg.blocks.size();
    constexpr size_t kPf = 16;
    // fp8 ingest compression (extension): bf16 pages stored at half size
    // with one scale per page; requires a 16-byte-divisible page.
    const bool quant = (msg.flags & kLocalFlagQuantFp8) != 0;
    if (quant && (page % 16 != 0)) return reply_local(c, ctx, INVALID_REQ);
    const size_t stored = quant ? page / 2 : page;

    // Phase A — dedup check only (shared stripe locks, prefetch-pipelined).
    // The authoritative first-write-wins decision happens at the insert pass
    // (phase D); a key that appears between A and D just wastes one block
    // copy whose memory is released after the kernel completes.
    std::vector<uint64_t> hashes(nb);
    for (size_t i = 0; i < nb; i++) hashes[i] = KvMap::hash_of(msg.blocks[i].first);
    std::vector<uint32_t> fresh;
    fresh.reserve(nb);
    // Bucket keys by stripe once; each stripe's lookups run under that
    // stripe's SHARED lock (~nb/16 keys per hold), so concurrent requests
    // on other stripes never wait.
    std::array<std::vector<uint32_t>, kStripes> by_stripe;
    for (size_t i = 0; i < nb; i++)
        by_stripe[stripe_of(hashes[i])].push_back(static_cast<uint32_t>(i));
    // Random start: concurrent requests sweeping stripes in the same
    // ascending order convoy on every lock in turn.
    size_t sweep0 = static_cast<size_t>(hashes[0] >> 32) % kStripes;
    double lw_us = 0, ul_us = 0;  // lock-wait vs under-lock (pdbg)
    for (size_t sk = 0; sk < kStripes; sk++) {
        size_t si = (sweep0 + sk) % kStripes;
        auto& list = by_stripe[si];
        if (list.empty()) continue;
        auto tl0 = pdbg ? std::chrono::steady_clock::now()
                        : std::chrono::steady_clock::time_point{};
        std::shared_lock<std::shared_mutex> lk(kv_[si].mu);
        auto tl1 = pdbg ? std::chrono::steady_clock::now()
                        : std::chrono::steady_clock::time_point{};
        auto& m = kv_[si].map;
        size_t ln = list.size();
        for (size_t i = 0; i < std::min(kPf, ln); i++) m.prefetch(hashes[list[i]]);
        for (size_t i = 0; i < ln; i++) {
            if (i + kPf < ln) m.prefetch(hashes[list[i + kPf]]);
            uint32_t gi = list[i];
            Ref<BlockEntry>* v = m.find_hashed(msg.blocks[gi].first, hashes[gi]);
            if (!v || expired(v->get())) fresh.push_back(gi);
        }
        if (pdbg) {
            auto tl2 = std::chrono::steady_clock::now();
            lw_us += std::chrono::duration<double, std::micro>(tl1 - tl0).count();
            ul_us += std::chrono::duration<double, std::micro>(tl2 - tl1).count();
        }
    }
    if (pdbg && nb > 64)
        fprintf(stderr, "[pdbg2] dedup lockwait=%.0f underlock=%.0f\n", lw_us, ul_us);
    auto p1 = std::chrono::steady_clock::now();

    n_writes_.fetch_add(1);
    size_t n_fresh = fresh.size();
    bytes_in_.fetch_add(n_fresh * page);
    bool sync_resp = (msg.flags & kLocalFlagSyncResponse) != 0;
    if (n_fresh == 0) {
        // everything was a duplicate — nothing to copy
        if (ctx.shm && !sync_resp) return;  // ring async writes are unacked
        return reply_local(c, ctx, sync_resp ? FINISH : TASK_ACCEPTED);
    }

    // Phase B — batched allocation (shard allocator lock only).
    std::vector<std::pair<void*, int>> slots;
    slots.reserve(n_fresh);
    auto try_alloc = [&] {
        slots.clear();
        return shard->allocate(stored, n_fresh,
                               [&](void* p, int idx) { slots.push_back({p, idx}); });
    };
    bool alloc_ok = try_alloc();
    // Retry loop: with concurrent writers a single evict+retry can lose its
    // freed blocks to a racing allocation; keep evicting until the
    // allocation lands or eviction runs dry.
    for (int attempt = 0; !alloc_ok && opt_.auto_evict && attempt < 4; attempt++) {
        if (evict_lru(shard, page * n_fresh * 2) == 0) break;
        alloc_ok = try_alloc();
    }
    if (!alloc_ok) return reply_local(c, ctx, OUT_OF_MEMORY);
    auto p2 = std::chrono::steady_clock::now();

    // Phase C — create entries, build the copy job and LAUNCH it; the index
    // insert pass (D) then runs while the kernel is in flight.
    uint64_t t = tick();
    SlabBatch slab_batch(n_fresh);
    auto entries = std::make_shared<std::vector<Ref<BlockEntry>>>();
    entries->reserve(n_fresh);
    Shard::CopyJob job;
    job.bytes_per_block = page;
    job.src.reserve(n_fresh);
    job.dst.reserve(n_fresh);
    auto scales_out =
        quant ? std::make_shared<std::vector<float>>(n_fresh, 1.f) : nullptr;
    if (quant) {
        job.xform = Shard::CopyJob::Xform::kQuantBf16Fp8;
        job.scales_out = scales_out;
    }
    for (size_t i = 0; i < n_fresh; i++) {
        auto* e = slab_batch.make();
        e->ptr = slots[i].first;
        e->size = stored;
        e->pool_idx = slots[i].second;
        e->shard = shard;
        e->committed = false;
        e->fp8 = quant;
        e->born_sec = now_sec();
        e->last_access.store(t, std::memory_order_relaxed);
        entries->emplace_back(e);
        job.src.push_back(
            reinterpret_cast<uint64_t>(client_ptr + msg.blocks[fresh[i]].second));
        job.dst.push_back(reinterpret_cast<uint64_t>(slots[i].first));
    }
    maybe_extend(shard);

    c->remain.fetch_add(1);
    c->ref();
    static const bool sdbg = getenv("IFS_SERVER_DEBUG") != nullptr;
    auto t_start = std::chrono::steady_clock::now();
    // won[i] is written in phase D after the kernel is submitted. The commit
    // must see phase D's result, so it runs on whichever side finishes LAST:
    // both the copy-done callback and the end of phase D bump `arrivals`,
    // and the second arriver performs the commit + reply. Round 1 used a
    // mutex held across submit + phase D instead, which made completion
    // threads BLOCK on handlers that were themselves waiting for copy
    // slots — the only slot-freers waiting on slot-waiters (measured 94 ms
    // slot waits at 64 saturation clients).
    auto won = std::make_shared<std::vector<uint8_t>>(n_fresh, 0);
    auto copy_ok = std::make_shared<std::atomic<bool>>(false);
    auto arrivals = std::make_shared<std::atomic<int>>(0);
    auto fin = [this, c, entries, won, sync_resp, ctx, scales_out, copy_ok] {
        bool ok = copy_ok->load(std::memory_order_acquire);
        if (ok) {
            for (size_t i = 0; i < entries->size(); i++)
                if ((*won)[i]) {
                    if (scales_out) (*entries)[i]->scale = (*scales_out)[i];
                    (*entries)[i]->committed = true;
                }
        } else {
            // Copy failed: drop the keys. Always from the owner loop —
            // erase_entries takes the kv lock exclusively, and compact()
            // waits on completion-thread futures WHILE holding that
            // lock, so taking it on a completion thread could deadlock.
            c->ref();
            c->owner->post([this, c, entries, won] {
                std::vector<Ref<BlockEntry>> winners;
                for (size_t i = 0; i < entries->size(); i++)
                    if ((*won)[i]) winners.push_back((*entries)[i]);
                erase_entries(winners);
                c->unref();
            });
        }
        if (sync_resp) reply_local(c, ctx, ok ? FINISH : INTERNAL_ERROR);
        finish_task(c, /*on_owner=*/!ctx.shm);
    };
    job.done = [this, c, entries, fin, ctx, copy_ok, arrivals, t_start](bool ok) {
        if (sdbg && entries->size() > 64) {
            auto us = std::chrono::duration<double, std::micro>(
                          std::chrono::steady_clock::now() - t_start)
                          .count();
            fprintf(stderr, "[sdbg] write n=%zu submit->complete=%.0fus\n", entries->size(),
                    us);
        }
        copy_ok->store(ok, std::memory_order_release);
        if (ctx.shm) {
            // Never block the completion thread: if phase D has not ended,
            // its own arrival runs fin.
            if (arrivals->fetch_add(1, std::memory_order_acq_rel) == 1) fin();
        } else {
            // Socket path: both arrivals happen on the owner loop thread
            // (this post + the handler itself), preserving reply ordering.
            c->owner->post([fin, arrivals] {
                if (arrivals->fetch_add(1, std::memory_order_acq_rel) == 1) fin();
            });
        }
    };
    // Async write: respond before submitting so the client's next request
    // overlaps the kernel. Sync-response write (flags&1): one round trip,
    // response sent on completion instead. Ring async writes are unacked.
    if (!sync_resp && !ctx.shm) send_status(c, TASK_ACCEPTED);
    bool submitted = shard->submit_copy(std::move(job));
    auto p2b = std::chrono::steady_clock::now();

    // Phase D — insert pass, overlapped with the in-flight kernel. Losers
    // (a racing writer inserted the key first) keep their block alive until
    // the copy completes, then release it; they are never committed.
    {
        std::array<std::vector<uint32_t>, kStripes> ins_by_stripe;  // pos in fresh
        for (uint32_t p = 0; p < n_fresh; p++)
            ins_by_stripe[stripe_of(hashes[fresh[p]])].push_back(p);
        size_t ins0 = static_cast<size_t>(hashes[fresh[0]] >> 32) % kStripes;
        double ilw_us = 0, iul_us = 0;
        for (size_t sk = 0; sk < kStripes; sk++) {
            size_t si = (ins0 + sk) % kStripes;
            auto& list = ins_by_stripe[si];
            if (list.empty()) continue;
            auto tl0 = pdbg ? std::chrono::steady_clock::now()
                            : std::chrono::steady_clock::time_point{};
            std::lock_guard<std::shared_mutex> lk(kv_[si].mu);
            auto tl1 = pdbg ? std::chrono::steady_clock::now()
                            : std::chrono::steady_clock::time_point{};
            struct UlAcc {  // accumulate under-lock time at scope exit
                bool on;
                std::chrono::steady_clock::time_point t1;
                double *lw, *ul;
                std::chrono::steady_clock::time_point t0;
                ~UlAcc() {
                    if (!on) return;
                    auto t2 = std::chrono::steady_clock::now();
                    *lw += std::chrono::duration<double, std::micro>(t1 - t0).count();
                    *ul += std::chrono::duration<double, std::micro>(t2 - t1).count();
                }
            } acc{pdbg, tl1, &ilw_us, &iul_us, tl0};
            auto& m = kv_[si].map;
            size_t ln = list.size();
            for (size_t i = 0; i < std::min(kPf, ln); i++) m.prefetch(hashes[fresh[list[i]]]);
            for (size_t i = 0; i < ln; i++) {
                if (i + kPf < ln) m.prefetch(hashes[fresh[list[i + kPf]]]);
                uint32_t p = list[i];
                bool inserted = false;
                Ref<BlockEntry>* slot2 = m.emplace_hashed(msg.blocks[fresh[p]].first,
                                                          hashes[fresh[p]], (*entries)[p],
                                                          &inserted);
                if (!inserted && slot2 && expired(slot2->get())) {
                    *slot2 = (*entries)[p];  // expired loser: replace in place
                    inserted = true;
                }
                (*won)[p] = inserted ? 1 : 0;
            }
        }
        if (pdbg && n_fresh > 64)
            fprintf(stderr, "[pdbg2] insert lockwait=%.0f underlock=%.0f\n", ilw_us, iul_us);
    }
    // Phase D done: second arriver commits (the copy may already be done).
    if (submitted && arrivals->fetch_add(1, std::memory_order_acq_rel) == 1) fin();
    auto p3 = std::chrono::steady_clock::now();
    if (pdbg && n_fresh > 64) {
        auto us = [](auto a, auto b) {
            return std::chrono::duration<double, std::micro>(b - a).count();
        };
        fprintf(stderr, "[pdbg] dedup=%.0f alloc=%.0f submit=%.0f insert=%.0f\n", us(p0, p1),
                us(p1, p2), us(p2, p2b), us(p2b, p3));
    }
    if (!submitted) {
        // The job never launched (e.g. a transform on a CPU shard): the done
        // callback will not fire, so answer sync-response writers here.
        if (sync_resp) reply_local(c, ctx, INTERNAL_ERROR);
        finish_task(c, /*on_owner=*/!ctx.shm);
    }
}

void Server::op_local_read(Conn* c, const LocalView& msg, const ReqCtx& ctx) {
    if (!gpu::available()) return reply_local(c, ctx, SYSTEM_ERROR);
    if (msg.ipc_len != gpu::kIpcHandleSize || msg.block_size <= 0)
        return reply_local(c, ctx, INVALID_REQ);
    IpcPinGuard pin(c);
    void* base = resolve_client_base(c, msg);
    if (!base) return reply_local(c, ctx, INTERNAL_ERROR);
    uint8_t* client_ptr = static_cast<uint8_t*>(base) + msg.base_offset;
    size_t page = static_cast<size_t>(msg.block_size);

    auto tr0 = std::chrono::steady_clock::now();
    // Group blocks by the shard whose GPU EXECUTES the copy. For same-GPU
    // hits that is trivially the owner. For cross-shard hits the choice is
    // deliberate: default OWNER-side (push) — the owner's
    // kernel reads its local HBM and writes the reader's memory over xGMI.
    // xGMI writes are posted (fire-and-forget per link), while remote reads
    // are request/response round trips; and since a multi-shard read groups
    // per owner, N owners push concurrently over N distinct xGMI links into
    // the reader (each GPU has 7 links x ~153 GB/s — per-link bound, so the
    // aggregate scales with the number of pushing GPUs, up to ~1 TB/s into
    // one reader). Reader-side execution (pull) would serialize the same
    // traffic through one GPU's remote-read path and its own CUs.
    // IFS_CROSS_COPY=reader flips to pull for topologies where the owner
    // GPUs are compute-saturated; it requires a shard on the reader's GPU.
    // fp8-compressed entries go into separate dequantizing jobs.
    static const bool cross_pull = [] {
        const char* v = getenv("IFS_CROSS_COPY");
        return v && std::string(v) == "reader";
    }();
    Shard* reader_shard = nullptr;
    if (cross_pull) {
        Shard* s = shard_for_device(msg.device);
        if (s->device() == msg.device) reader_shard = s;
    }
    auto exec_shard = [&](Shard* owner) {
        return (reader_shard && owner->device() != msg.device) ? reader_shard : owner;
    };
    std::map<Shard*, Shard::CopyJob> jobs;
    std::map<Shard*, Shard::CopyJob> qjobs;
    auto held = std::make_shared<std::vector<Ref<BlockEntry>>>();
    held->reserve(msg.blocks.size());
    uint64_t read_tick = tick();
    {
        size_t nb2 = msg.blocks.size();
        std::vector<uint64_t> hashes(nb2);
        for (size_t i = 0; i < nb2; i++) hashes[i] = KvMap::hash_of(msg.blocks[i].first);
        constexpr size_t kPf = 16;
        std::array<std::vector<uint32_t>, kStripes> by_stripe;
        for (size_t i = 0; i < nb2; i++)
            by_stripe[stripe_of(hashes[i])].push_back(static_cast<uint32_t>(i));
        for (size_t si = 0; si < kStripes; si++) {
            auto& list = by_stripe[si];
            if (list.empty()) continue;
            std::shared_lock<std::shared_mutex> lk(kv_[si].mu);
            auto& m = kv_[si].map;
            size_t ln = list.size();
            for (size_t i = 0; i < std::min(kPf, ln); i++) m.prefetch(hashes[list[i]]);
            for (size_t i = 0; i < ln; i++) {
                if (i + kPf < ln) m.prefetch(hashes[list[i + kPf]]);
                auto& b = msg.blocks[list[i]];
                Ref<BlockEntry>* v = m.find_hashed(b.first, hashes[list[i]]);
                if (!v || !(*v)->committed || expired(v->get())) {
                    return reply_local(c, ctx, KEY_NOT_FOUND);
                }
                BlockEntry* e = v->get();
                e->last_access.store(read_tick, std::memory_order_relaxed);
                if (e->fp8) {
                    if (e->size * 2 != page) return reply_local(c, ctx, INVALID_REQ);
                    auto& qj = qjobs[exec_shard(e->shard)];
                    qj.bytes_per_block = page;
                    qj.xform = Shard::CopyJob::Xform::kDequantFp8Bf16;
                    qj.scales_in.push_back(e->scale);
                    qj.src.push_back(reinterpret_cast<uint64_t>(e->ptr));
                    qj.dst.push_back(reinterpret_cast<uint64_t>(client_ptr + b.second));
                } else {
                    auto& job = jobs[exec_shard(e->shard)];
                    job.bytes_per_block = page;
                    job.src.push_back(reinterpret_cast<uint64_t>(e->ptr));
                    job.dst.push_back(reinterpret_cast<uint64_t>(client_ptr + b.second));
                }
                held->push_back(*v);
            }
        }
    }
    static const bool rdbg2 = getenv("IFS_SERVER_DEBUG") != nullptr;
    auto tcol = std::chrono::steady_clock::now();
    n_reads_.fetch_add(1);
    bytes_out_.fetch_add(msg.blocks.size() * page);
    bool sync_resp = (msg.flags & kLocalFlagSyncResponse) != 0;
    if (jobs.empty() && qjobs.empty())
        return reply_local(c, ctx, sync_resp ? FINISH : TASK_ACCEPTED);

    c->remain.fetch_add(1);
    c->ref();
    auto pending =
        std::make_shared<std::atomic<int>>(static_cast<int>(jobs.size() + qjobs.size()));
    auto all_ok = std::make_shared<std::atomic<bool>>(true);
    auto submit_one = [&](Shard* shard, Shard::CopyJob& job) -> bool {
        Shard::CopyJob j = std::move(job);
        static const bool sdbg2 = getenv("IFS_SERVER_DEBUG") != nullptr;
        auto t_start2 = std::chrono::steady_clock::now();
        j.done = [this, c, held, pending, all_ok, sync_resp, ctx, t_start2](bool ok) {
            if (sdbg2 && held->size() > 64) {
                auto us = std::chrono::duration<double, std::micro>(
                              std::chrono::steady_clock::now() - t_start2)
                              .count();
                fprintf(stderr, "[sdbg] read n=%zu submit->complete=%.0fus\n", held->size(), us);
            }
            if (!ok) all_ok->store(false);
            if (pending->fetch_sub(1) == 1) {
                if (ctx.shm) {
                    // reply straight from the completion thread: the ring
                    // write needs no loop-thread affinity.
                    if (sync_resp) reply_local(c, ctx, all_ok->load() ? FINISH : INTERNAL_ERROR);
                    finish_task(c, /*on_owner=*/false);
                } else {
                    c->owner->post([this, c, held, all_ok, sync_resp] {
                        if (sync_resp)
                            send_status(c, all_ok->load() ? FINISH : INTERNAL_ERROR);
                        finish_task(c, /*on_owner=*/true);
                    });
                }
            }
        };
        if (!shard->submit_copy(std::move(j))) {
            all_ok->store(false);
            if (pending->fetch_sub(1) == 1) {
                finish_task(c, /*on_owner=*/!ctx.shm);
                reply_local(c, ctx, INTERNAL_ERROR);
                return false;  // error reply sent; stop submitting
            }
        }
        return true;
    };
    for (auto& [shard, job] : jobs)
        if (!submit_one(shard, job)) return;
    for (auto& [shard, job] : qjobs)
        if (!submit_one(shard, job)) return;
    if (rdbg2 && msg.blocks.size() > 256) {
        auto us = [](auto a, auto b) {
            return std::chrono::duration<double, std::micro>(b - a).count();
        };
        auto tsub = std::chrono::steady_clock::now();
        fprintf(stderr, "[rdbg] read n=%zu collect=%.0f submit=%.0f\n", msg.blocks.size(),
                us(tr0, tcol), us(tcol, tsub));
    }
    if (!sync_resp && !ctx.shm) send_status(c, TASK_ACCEPTED);
}

void Server::op_sync(Conn* c, const ReqCtx& ctx) {
    std::lock_guard<std::mutex> lk(c->sync_mu);
    if (c->remain.load() == 0) {
        reply_query(c, ctx, 0);  // FINISH + remain (infinistore.cpp:1070-1075)
    } else {
        c->sync_waiting = true;  // answered by finish_task when remain drains
        c->sync_ctx = ctx;
    }
}

// Owner loop thread (socket completions) or a completion thread (shm).
void Server::finish_task(Conn* c, bool on_owner) {
    if (c->remain.fetch_sub(1) == 1) {
        ReqCtx ctx;
        bool respond = false;
        {
            std::lock_guard<std::mutex> lk(c->sync_mu);
            if (c->sync_waiting) {
                c->sync_waiting = false;
                respond = true;
                ctx = c->sync_ctx;
            }
        }
        if (respond) {
            if (ctx.shm || on_owner) {
                reply_query(c, ctx, 0);
            } else {
                // socket reply from a completion thread: hop to the owner
                // loop (uv_write is not thread-safe).
                c->ref();
                c->owner->post([this, c, ctx] {
                    reply_query(c, ctx, 0);
                    c->unref();
                });
            }
        }
    }
    c->unref();
}

void Server::reply_local(Conn* c, const ReqCtx& ctx, int code) {
    if (ctx.shm && c->shm) return c->shm->push_resp(ctx.seq, code);
    send_status(c, code);
}

// Query-op responses (check_exist / match / sync): the reference frames
// these as FINISH + 4-byte value on the socket (libinfinistore.cpp:679-694,
// infinistore.cpp:1070-1107); the shm ring carries the value in its
// fixed-size status field instead (ring responses are always 32 bytes).
void Server::reply_query(Conn* c, const ReqCtx& ctx, int value) {
    if (ctx.shm && c->shm) return c->shm->push_resp(ctx.seq, value);
    send_status_raw(c, FINISH, &value, 4);
}

// ---- shared-memory ring transport -----------------------------------------
void Server::ShmPeer::push_resp(uint64_t seq, int status) {
    std::lock_guard<std::mutex> lk(resp_mu);
    shmring::RespRec r{};
    r.h.len = sizeof(r);
    r.h.op = 0;
    r.h.body_len = sizeof(r) - sizeof(r.h);
    r.h.seq = seq;
    r.status = status;
    uint64_t adv = 0;
    uint8_t* dst = seg.resp->claim(sizeof(r), &adv);
    if (!dst) {
        // Client stopped draining (crashed / gone): drop the response; the
        // socket EOF tears the conn down.
        WARN("shm resp ring full; dropping response seq=%llu",
             static_cast<unsigned long long>(seq));
        return;
    }
    memcpy(dst, &r, sizeof(r));
    seg.resp->publish(adv);
}

void Server::op_shm_setup(Conn* c, const std::vector<uint8_t>& body) {
    // Body: shm_open name of a client-created segment. Refuse when already
    // set up, name is implausible, or too many pollers are running.
    if (c->shm || body.empty() || body.size() > 100 || body[0] != '/')
        return send_status(c, INVALID_REQ);
    // Cap on ring pollers: sized for 8 ranks x (4 worker pairs + 1 parent)
    // plus 64 saturation clients with headroom (overflow conns gracefully
    // stay on the socket).
    if (shm_peers_.load() >= 160) return send_status(c, SYSTEM_ERROR);
    std::string name(reinterpret_cast<const char*>(body.data()), body.size());
    auto* p = new ShmPeer();
    p->srv = this;
    p->c = c;
    if (!shmring::open_segment(name, &p->seg)) {
        delete p;
        return send_status(c, INTERNAL_ERROR);
    }
    c->shm = p;
    shm_peers_.fetch_add(1);
    c->ref();  // held by the poller thread; released at poller exit
    p->th = std::thread([this, p] { shm_poll_main(p); });
    INFO("shm ring attached (%zu bytes)", p->seg.len);
    send_status(c, FINISH);
}

void Server::shm_teardown(Conn* c) {
    if (!c->shm) return;
    c->shm->stop.store(true, std::memory_order_release);
    if (c->shm->th.joinable()) c->shm->th.join();
    shm_peers_.fetch_sub(1);
    // The peer + mapping are freed in ~Conn: in-flight completion threads
    // may still write responses into the (now client-less) ring.
}

void Server::fast_worker_main() {
    for (;;) {
        FastWork w;
        {
            std::unique_lock<std::mutex> lk(fast_mu_);
            fast_cv_.wait(lk, [this] { return stop_requested_.load() || !fast_q_.empty(); });
            if (fast_q_.empty()) {
                if (stop_requested_.load()) return;
                continue;
            }
            w = std::move(fast_q_.front());
            fast_q_.pop_front();
        }
        run_fast(w);
    }
}

void Server::run_fast(FastWork& w) {
    ReqCtx ctx{w.seq, true};
    LocalView v;
    if (!parse_packed_local(w.body.data(), w.body.size(), &v)) {
        reply_local(w.c, ctx, INVALID_REQ);
    } else if (w.op == OP_W_FAST) {
        op_local_write(w.c, v, ctx);
    } else {
        op_local_read(w.c, v, ctx);
    }
    // Balance the poller's enqueue-time remain bump (+ answer a deferred
    // sync if this was the last in-flight unit).
    finish_task(w.c, /*on_owner=*/false);
}

void Server::shm_poll_main(ShmPeer* p) {
    Conn* c = p->c;
    auto last_work = std::chrono::steady_clock::now();
    while (!p->stop.load(std::memory_order_acquire)) {
        uint32_t len = 0;
        uint64_t skip = 0;
        bool corrupt = false;
        const uint8_t* rec = p->seg.req->peek(&len, &skip, &corrupt);
        if (corrupt) {
            // Provably invalid record header (hostile or broken client):
            // following it could read past the mapped segment. Stop polling
            // this ring for good; the conn's socket teardown reclaims it.
            ERROR("shm req ring corrupt (invalid record header); detaching poller");
            break;
        }
        if (!rec) {
            // Adaptive idle: spin while recently hot (sub-µs pickup), back
            // off to 50 µs sleeps when the ring has been quiet.
            if (std::chrono::steady_clock::now() - last_work > std::chrono::milliseconds(2)) {
                usleep(50);
            } else {
#if defined(__x86_64__)
                for (int i = 0; i < 64; i++) __builtin_ia32_pause();
#else
                std::this_thread::yield();
#endif
            }
            continue;
        }
        if (len == 0) {  // wrap marker
            p->seg.req->consume(skip);
            continue;
        }
        shmring::RecHdr h;
        memcpy(&h, rec, sizeof(h));
        const uint8_t* body = rec + sizeof(h);
        ReqCtx ctx{h.seq, true};
        auto t0 = std::chrono::steady_clock::now();
        char op = static_cast<char>(h.op);
        static const bool rdbg = getenv("IFS_SERVER_DEBUG") != nullptr;
        if (rdbg && h.body_len > 4096)
            fprintf(stderr, "[rdbg] op=%c pickup_wait=%uus\n", op,
                    shmring::mono_us() - h.t_push_us);
        if (h.body_len + sizeof(h) > len) {
            reply_local(c, ctx, INVALID_REQ);
        } else {
            // The record stays in the ring while the handler runs (views
            // point into it); consume() below frees the space.
            switch (op) {
                case OP_W_FAST:
                case OP_R_FAST: {
                    // Hand off to the fast-op pool so this conn's burst of
                    // async requests is handled in parallel (inline handling
                    // serialized each conn behind ~ms handlers). remain is
                    // bumped HERE so a later OP_SYNC on this ring (FIFO)
                    // can never miss the request; the worker balances it.
                    // Hand off only when it pays: many active ring clients
                    // (the pool exists to break per-conn serialization at
                    // saturation — with a handful of pipelined conns the
                    // extra hop just adds latency) and a body big enough
                    // that the handler dwarfs the wakeup (~10 µs vs a
                    // ~20 µs single-page round trip).
                    size_t qlen = 0;
                    bool pool = h.body_len >= 2048 && shm_peers_.load() >= 8 &&
                                !fast_workers_.empty();
                    if (pool) {
                        std::lock_guard<std::mutex> lk(fast_mu_);
                        qlen = fast_q_.size();
                    }
                    if (pool && qlen < 8192) {
                        FastWork w;
                        w.c = c;
                        w.op = op;
                        w.seq = h.seq;
                        w.body.assign(body, body + h.body_len);
                        c->remain.fetch_add(1);
                        c->ref();
                        {
                            std::lock_guard<std::mutex> lk(fast_mu_);
                            fast_q_.push_back(std::move(w));
                        }
                        fast_cv_.notify_one();
                        break;
                    }
                    LocalView v;  // overload fallback: handle inline
                    if (!parse_packed_local(body, h.body_len, &v))
                        reply_local(c, ctx, INVALID_REQ);
                    else if (op == OP_W_FAST)
                        op_local_write(c, v, ctx);
                    else
                        op_local_read(c, v, ctx);
                    break;
                }
                case OP_SYNC:
                    op_sync(c, ctx);
                    break;
                case OP_CHECK_EXIST:
                    op_check_exist(c, std::vector<uint8_t>(body, body + h.body_len), ctx);
                    break;
                case OP_GET_MATCH_LAST_IDX:
                    op_match_index(c, std::vector<uint8_t>(body, body + h.body_len), ctx);
                    break;
                case OP_DELETE:
                    op_delete(c, std::vector<uint8_t>(body, body + h.body_len), ctx);
                    break;
                default:
                    reply_local(c, ctx, INVALID_REQ);
            }
        }
        p->seg.req->consume(skip);
        last_work = std::chrono::steady_clock::now();
        auto us = std::chrono::duration_cast<std::chrono::microseconds>(last_work - t0).count();
        auto& st = op_stats_[static_cast<uint8_t>(op) & 127];
        st.count.fetch_add(1, std::memory_order_relaxed);
        st.total_us.fetch_add(static_cast<uint64_t>(us), std::memory_order_relaxed);
        uint64_t prev = st.max_us.load(std::memory_order_relaxed);
        while (static_cast<uint64_t>(us) > prev &&
               !st.max_us.compare_exchange_weak(prev, static_cast<uint64_t>(us))) {
        }
    }
    c->unref();
}

// ---- fabric negotiation + RDMA-semantics ops ------------------------------
void Server::op_exchange(Conn* c, const std::vector<uint8_t>& body) {
    // A body carrying ConnInfo requests the verbs fabric; succeed only when
    // rdma-core + an active NIC are present, else fall back to the TCP data
    // fabric ("TCPF" tag, payloads inline on this socket).
    if (body.size() >= sizeof(vf::ConnInfo)) {
        std::vector<uint8_t> reply;
        if (verbs_handshake(c, body, &reply)) {
            return send_status_payload(c, FINISH, reply.data(), reply.size());
        }
    }
    const char tag[4] = {'T', 'C', 'P', 'F'};
    send_status_payload(c, FINISH, reinterpret_cast<const uint8_t*>(tag), 4);
}

std::vector<RemoteBlockWire> Server::allocate_blocks(Conn* c,
                                                     const std::vector<std::string>& keys,
                                                     size_t page, int* status) {
    *status = FINISH;
    Shard* shard = shard_least_used();
    std::vector<RemoteBlockWire> blocks;
    blocks.reserve(keys.size());
    std::vector<std::tuple<std::string, uint64_t, uint64_t>> created;  // key, hash, ptr
    auto rollback = [&] {
        for (auto& [k, h, p] : created) {
            {
                auto& st = kv_[stripe_of(h)];
                std::lock_guard<std::shared_mutex> lk(st.mu);
                st.map.erase(k);
            }
            c->pending_rdma.erase(p);  // drop the ref so the block frees now
        }
    };
    for (auto& key : keys) {
        uint64_t h = KvMap::hash_of(key);
        auto& st = kv_[stripe_of(h)];
        {
            std::shared_lock<std::shared_mutex> sl(st.mu);
            Ref<BlockEntry>* v0 = st.map.find_hashed(key, h);
            if (v0 != nullptr && !expired(v0->get())) {
                blocks.push_back({0, 0, 0});  // FAKE block: dup key, client skips
                continue;
            }
        }
        void* ptr = nullptr;
        int pool_idx = -1;
        auto try_alloc = [&] {
            return shard->allocate(page, 1, [&](void* p, int idx) {
                ptr = p;
                pool_idx = idx;
            });
        };
        bool ok = try_alloc();
        for (int attempt = 0; !ok && opt_.auto_evict && attempt < 4; attempt++) {
            if (evict_lru(shard, page * 4) == 0) break;
            ok = try_alloc();
        }
        if (!ok) {
            rollback();
            *status = OUT_OF_MEMORY;
            return {};
        }
        auto* e = new BlockEntry();
        e->ptr = ptr;
        e->size = page;
        e->pool_idx = pool_idx;
        e->shard = shard;
        e->born_sec = now_sec();
        e->last_access.store(tick(), std::memory_order_relaxed);
        Ref<BlockEntry> ref(e);
        bool ins = false;
        {
            std::lock_guard<std::shared_mutex> lk(st.mu);
            Ref<BlockEntry>* slot2 = st.map.emplace_hashed(key, h, ref, &ins);
            if (!ins && slot2 && expired(slot2->get())) {
                *slot2 = ref;  // expired entry: replace
                ins = true;
            }
        }
        if (!ins) {  // raced with another writer: first write wins
            blocks.push_back({0, 0, 0});
            continue;  // `ref` releases and frees the block via ~BlockEntry
        }
        // rkey: on the verbs fabric the client posts one-sided RDMA_WRITEs
        // with this key, so it MUST be the registered MR's real rkey for the
        // arena holding the block (reference returns get_rkey(pool_idx) per
        // block, infinistore.cpp:382-396 via mempool.h:56-84). On the TCP
        // fabric the field is unused by the client (addresses key the
        // pending map); an opaque nonzero marker keeps it distinguishable
        // from the FAKE dup sentinel {0,0}.
        uint32_t rkey = static_cast<uint32_t>(shard->device() + 2);
        if (c->verbs) {
            vf::MrInfo mr;
            if (!vdrv_ || !vdrv_->lookup_region(ptr, &mr)) {
                ERROR("allocate: pool arena %p has no MR on the verbs fabric", ptr);
                {
                    std::lock_guard<std::shared_mutex> lk(st.mu);
                    st.map.erase(key);
                }
                rollback();
                *status = SYSTEM_ERROR;
                return {};
            }
            rkey = mr.rkey;
        }
        created.push_back({key, h, reinterpret_cast<uint64_t>(ptr)});
        c->pending_rdma.emplace(reinterpret_cast<uint64_t>(ptr), ref);
        blocks.push_back({rkey, 0, reinterpret_cast<uint64_t>(ptr)});
    }
    maybe_extend(shard);
    return blocks;
}

void Server::commit_addrs(Conn* c, const std::vector<uint64_t>& addrs) {
    for (uint64_t addr : addrs) {
        auto it = c->pending_rdma.find(addr);
        if (it == c->pending_rdma.end()) continue;
        it->second->committed = true;
        c->pending_rdma.erase(it);
    }
}

bool Server::collect_read_entries(const std::vector<std::string>& keys,
                                  std::vector<Ref<BlockEntry>>* out) {
    out->reserve(keys.size());
    uint64_t t = tick();
    for (auto& key : keys) {
        uint64_t h = KvMap::hash_of(key);
        auto& st = kv_[stripe_of(h)];
        std::shared_lock<std::shared_mutex> lk(st.mu);
        Ref<BlockEntry>* v = st.map.find_hashed(key, h);
        if (!v || !(*v)->committed || expired(v->get())) return false;
        // fp8-compressed entries are a local-GPU-path feature: the TCP/verbs
        // fabric moves raw bytes and cannot dequantize on the way out.
        if ((*v)->fp8) return false;
        (*v)->last_access.store(t, std::memory_order_relaxed);
        out->push_back(*v);
    }
    return true;
}

void Server::op_allocate(Conn* c, const RemoteMetaMsg& msg) {
    if (msg.block_size <= 0 || msg.keys.empty()) return send_status(c, INVALID_REQ);
    int status = FINISH;
    auto blocks = allocate_blocks(c, msg.keys, static_cast<size_t>(msg.block_size), &status);
    if (status != FINISH) return send_status(c, status);
    auto payload = build_allocate_response(blocks);
    send_status_payload(c, FINISH, payload.data(), payload.size());
}

void Server::op_tcp_put(Conn* c, std::vector<uint8_t> body) {
    // Layout: [u32 n][u32 block_size][u64 addr x n][payload n*block_size]
    if (body.size() < 8) return send_status(c, INVALID_REQ);
    uint32_t n, bs;
    memcpy(&n, body.data(), 4);
    memcpy(&bs, body.data() + 4, 4);
    size_t need = 8 + static_cast<size_t>(n) * 8 + static_cast<size_t>(n) * bs;
    if (n == 0 || bs == 0 || body.size() < need) return send_status(c, INVALID_REQ);
    const uint64_t* addrs = reinterpret_cast<const uint64_t*>(body.data() + 8);
    size_t payload_off = 8 + static_cast<size_t>(n) * 8;

    // Validate addresses and group by owning shard; copies then run on the
    // shards' fabric worker threads (the IO loop never blocks on memcpy).
    std::map<Shard*, Shard::FabricJob> jobs;
    std::vector<Ref<BlockEntry>> held;
    held.reserve(n);
    for (uint32_t i = 0; i < n; i++) {
        auto it = c->pending_rdma.find(addrs[i]);
        if (it == c->pending_rdma.end()) {
            WARN("tcp_put: unknown addr %llx", (unsigned long long)addrs[i]);
            return send_status(c, INVALID_REQ);
        }
        BlockEntry* e = it->second.get();
        if (e->size < bs) return send_status(c, INVALID_REQ);
        auto& j = jobs[e->shard];
        j.is_put = true;
        j.bytes_per_block = bs;
        j.block_ptrs.push_back(reinterpret_cast<uint64_t>(e->ptr));
        j.host_offsets.push_back(payload_off + static_cast<size_t>(i) * bs);
        held.push_back(it->second);
    }
    n_put_.fetch_add(1);
    bytes_in_.fetch_add(static_cast<size_t>(n) * bs);

    auto shared_body = std::make_shared<std::vector<uint8_t>>(std::move(body));
    auto pending = std::make_shared<std::atomic<int>>(static_cast<int>(jobs.size()));
    auto all_ok = std::make_shared<std::atomic<bool>>(true);
    auto held_sp = std::make_shared<std::vector<Ref<BlockEntry>>>(std::move(held));
    c->fabric_inflight.fetch_add(1);
    c->ref();
    for (auto& [shard, job] : jobs) {
        Shard::FabricJob j = std::move(job);
        j.host = shared_body;
        j.done = [this, c, pending, all_ok, held_sp](bool ok) {
            if (!ok) all_ok->store(false);
            if (pending->fetch_sub(1) == 1) {
                c->owner->post([this, c, all_ok, held_sp] {
                    send_status(c, all_ok->load() ? TASK_ACCEPTED : INTERNAL_ERROR);
                    if (c->fabric_inflight.fetch_sub(1) == 1) {
                        // Drain commits that arrived while copies were in flight.
                        for (auto& addrs2 : c->deferred_commits) {
                            commit_addrs(c, addrs2);
                            send_status(c, FINISH);
                        }
                        c->deferred_commits.clear();
                    }
                    c->unref();
                });
            }
        };
        shard->submit_fabric(std::move(j));
    }
}

void Server::op_commit(Conn* c, const RemoteMetaMsg& msg) {
    if (c->fabric_inflight.load() > 0) {
        // Puts still copying on the fabric workers: commit (and its ACK)
        // must wait so readers never see committed-but-unwritten blocks.
        c->deferred_commits.push_back(msg.remote_addrs);
        return;
    }
    commit_addrs(c, msg.remote_addrs);
    send_status(c, FINISH);
}

void Server::op_tcp_get(Conn* c, const RemoteMetaMsg& msg) {
    if (msg.block_size <= 0 || msg.keys.empty()) return send_status(c, INVALID_REQ);
    size_t page = static_cast<size_t>(msg.block_size);
    std::vector<Ref<BlockEntry>> entries;
    if (!collect_read_entries(msg.keys, &entries)) return send_status(c, KEY_NOT_FOUND);
    // The framed response (status + len + blocks) is ONE buffer, filled by
    // the shards' fabric workers; the response is sent on completion.
    // Allocated UNINITIALIZED: value-initializing a 256 MB vector costs
    // ~100 ms before any data moves.
    size_t total = entries.size() * page;
    std::shared_ptr<uint8_t[]> resp(new uint8_t[8 + total]);
    int code = FINISH;
    uint32_t len32 = static_cast<uint32_t>(total);
    memcpy(resp.get(), &code, 4);
    memcpy(resp.get() + 4, &len32, 4);
    std::map<Shard*, Shard::FabricJob> jobs;
    for (size_t i = 0; i < entries.size(); i++) {
        BlockEntry* e = entries[i].get();
        auto& j = jobs[e->shard];
        j.is_put = false;
        j.bytes_per_block = page;
        j.block_ptrs.push_back(reinterpret_cast<uint64_t>(e->ptr));
        j.host_offsets.push_back(8 + i * page);
    }
    n_get_.fetch_add(1);
    bytes_out_.fetch_add(total);
    auto pending = std::make_shared<std::atomic<int>>(static_cast<int>(jobs.size()));
    auto all_ok = std::make_shared<std::atomic<bool>>(true);
    auto held_sp = std::make_shared<std::vector<Ref<BlockEntry>>>(std::move(entries));
    c->ref();
    size_t resp_len = 8 + total;

    // Single-shard big responses STREAM: the job is split into segments and
    // each segment is written to the socket as its copy lands, overlapping
    // the pool→host copy with the send (they were strictly sequential —
    // 256 MB paid copy + send back to back). Payload offsets are ascending
    // within the single job and the fabric worker runs segments FIFO, so
    // in-order streaming holds. Multi-shard responses interleave offsets
    // and keep the single-shot path.
    constexpr size_t kSegBytes = 16u << 20;
    if (jobs.size() == 1 && total > 2 * kSegBytes) {
        auto& [shard, whole] = *jobs.begin();
        size_t nb = whole.block_ptrs.size();
        size_t per_seg = std::max<size_t>(1, kSegBytes / page);
        size_t n_segs = (nb + per_seg - 1) / per_seg;
        auto segs_left = std::make_shared<std::atomic<int>>(static_cast<int>(n_segs));
        for (size_t off = 0; off < nb; off += per_seg) {
            size_t take = std::min(per_seg, nb - off);
            Shard::FabricJob j;
            j.is_put = false;
            j.bytes_per_block = page;
            j.raw_host = resp;
            j.block_ptrs.assign(whole.block_ptrs.begin() + static_cast<long>(off),
                                whole.block_ptrs.begin() + static_cast<long>(off + take));
            j.host_offsets.assign(whole.host_offsets.begin() + static_cast<long>(off),
                                  whole.host_offsets.begin() + static_cast<long>(off + take));
            bool first = off == 0;
            size_t seg_start = first ? 0 : 8 + off * page;  // frame rides seg 0
            size_t seg_len = (first ? 8 : 0) + take * page;
            j.done = [this, c, resp, segs_left, all_ok, held_sp, seg_start,
                      seg_len](bool ok) {
                if (!ok) all_ok->store(false);
                bool last = segs_left->fetch_sub(1) == 1;
                c->ref();
                c->owner->post([this, c, resp, all_ok, seg_start, seg_len, last] {
                    if (all_ok->load()) {
                        std::shared_ptr<uint8_t[]> view(resp, resp.get() + seg_start);
                        send_raw(c, std::move(view), seg_len);
                    } else if (last) {
                        // some earlier segments may already be on the wire;
                        // the client's length-checked read will fail cleanly
                        conn_close(c);
                    }
                    c->unref();
                    if (last) c->unref();  // the op's own ref
                });
            };
            shard->submit_fabric(std::move(j));
        }
        return;
    }

    for (auto& [shard, job] : jobs) {
        Shard::FabricJob j = std::move(job);
        j.raw_host = resp;
        j.done = [this, c, resp, resp_len, pending, all_ok, held_sp](bool ok) {
            if (!ok) all_ok->store(false);
            if (pending->fetch_sub(1) == 1) {
                c->owner->post([this, c, resp, resp_len, all_ok] {
                    if (all_ok->load())
                        send_raw(c, resp, resp_len);
                    else
                        send_status(c, INTERNAL_ERROR);
                    c->unref();
                });
            }
        };
        shard->submit_fabric(std::move(j));
    }
}

// ---- queries ---------------------------------------------------------------
void Server::op_check_exist(Conn* c, const std::vector<uint8_t>& body, const ReqCtx& ctx) {
    // body: raw key bytes (reference framing — body_size IS the key length,
    // libinfinistore.cpp:659-671); response FINISH + int, 0 = exists
    // (infinistore.cpp:1078-1090).
    if (body.empty()) return reply_local(c, ctx, INVALID_REQ);
    std::string key(reinterpret_cast<const char*>(body.data()), body.size());
    bool exists;
    {
        uint64_t h = KvMap::hash_of(key);
        auto& st = kv_[stripe_of(h)];
        std::shared_lock<std::shared_mutex> lk(st.mu);
        Ref<BlockEntry>* v = st.map.find_hashed(key, h);
        exists = v && (*v)->committed && !expired(v->get());
    }
    reply_query(c, ctx, exists ? 0 : 1);
}

void Server::op_match_index(Conn* c, const std::vector<uint8_t>& body, const ReqCtx& ctx) {
    std::vector<std::string> keys;
    if (!parse_match_request(body.data(), body.size(), &keys) || keys.empty())
        return reply_query(c, ctx, -1);
    // Binary search for the last present index, assuming the prefix property
    // (keys[0..i] present iff i <= match). Requires committed entries —
    // divergence from the reference, which counts uncommitted keys as
    // present (infinistore.cpp:1097); an uncommitted key cannot be read, so
    // reporting it as a hit would make the subsequent read_cache fail.
    auto present = [&](size_t i) {
        uint64_t h = KvMap::hash_of(keys[i]);
        auto& st = kv_[stripe_of(h)];
        std::shared_lock<std::shared_mutex> lk(st.mu);
        Ref<BlockEntry>* v = st.map.find_hashed(keys[i], h);
        return v && (*v)->committed && !expired(v->get());
    };
    long left = 0, right = static_cast<long>(keys.size());
    while (left < right) {
        long mid = (left + right) / 2;
        if (present(static_cast<size_t>(mid)))
            left = mid + 1;
        else
            right = mid;
    }
    reply_query(c, ctx, static_cast<int>(left - 1));
}

void Server::op_delete(Conn* c, const std::vector<uint8_t>& body, const ReqCtx& ctx) {
    std::vector<std::string> keys;
    if (!parse_match_request(body.data(), body.size(), &keys))
        return reply_local(c, ctx, INVALID_REQ);
    int n = 0;
    {
        std::array<std::vector<uint32_t>, kStripes> by_stripe;
        std::vector<uint64_t> hashes(keys.size());
        for (size_t i = 0; i < keys.size(); i++) {
            hashes[i] = KvMap::hash_of(keys[i]);
            by_stripe[stripe_of(hashes[i])].push_back(static_cast<uint32_t>(i));
        }
        constexpr size_t kPf = 16;
        // Move the refs out under the lock, destroy them after unlocking:
        // entry destruction frees pool blocks (shard alloc lock) + slab
        // memory, which must never run under an exclusive stripe lock (it
        // blocked every op on the stripe for ms during delete sweeps).
        std::vector<Ref<BlockEntry>> dead;
        dead.reserve(keys.size());
        // Sole-owner entries surrender their block here and the blocks free
        // in ONE allocator-lock hold per shard — per-block frees convoyed
        // against concurrent batch allocations (1.3 s per 10k-key delete).
        std::map<Shard*, std::vector<Shard::BlockFree>> by_shard;
        for (size_t si = 0; si < kStripes; si++) {
            auto& list = by_stripe[si];
            if (list.empty()) continue;
            {
                std::lock_guard<std::shared_mutex> lk(kv_[si].mu);
                auto& m = kv_[si].map;
                for (size_t i = 0; i < std::min(kPf, list.size()); i++)
                    m.prefetch(hashes[list[i]]);
                for (size_t i = 0; i < list.size(); i++) {
                    if (i + kPf < list.size()) m.prefetch(hashes[list[i + kPf]]);
                    Ref<BlockEntry> ref;
                    if (m.extract(keys[list[i]], &ref)) {
                        n++;
                        dead.emplace_back(std::move(ref));
                    }
                }
            }
            for (auto& ref : dead) {
                BlockEntry* e = ref.get();
                if (e->ref_count() == 1 && e->shard && e->ptr) {
                    by_shard[e->shard].push_back({e->ptr, e->size, e->pool_idx});
                    e->shard = nullptr;  // dtor skips the per-block free
                    e->ptr = nullptr;
                }
            }
            dead.clear();  // destructors run lock-free (per-stripe batch)
        }
        for (auto& [shard, frees] : by_shard) shard->deallocate_bulk(frees);
    }
    reply_local(c, ctx, n);
}

// ---------------------------------------------------------------------------
// Management plane
// ---------------------------------------------------------------------------
std::pair<size_t, size_t> Server::compact() {
    if (!running_.load()) return {0, 0};
    // Defragmentation in three phases per shard. The stripe locks are NOT
    // held across the copy wait: a write handler holds its job's commit
    // mutex while its insert pass takes stripe locks, and the completion
    // thread that fulfills our copy futures can be blocked on that same
    // commit mutex — holding the stripes here closed that into a deadlock
    // (found by the mixed fp8/eviction soak).
    //  1. snapshot (all stripe locks, in order): pick committed idle
    //     entries, take a Ref on each so neither delete nor eviction can
    //     free the old block mid-copy.
    //  2. copy (no locks): batched kernel moves into the planned slots.
    //  3. swap (all stripe locks): re-check each entry is still present and
    //     still idle (a reader that started during the copy captured the
    //     OLD pointer and holds a ref — skip those), then flip ptr/pool and
    //     free the old slot. Unapplied moves free their new slot instead.
    size_t moved = 0, bytes = 0;
    for (auto& shard_up : shards_) {
        Shard* shard = shard_up.get();
        std::vector<Ref<BlockEntry>> held;
        std::vector<std::pair<void*, size_t>> movable;
        {
            std::vector<std::unique_lock<std::shared_mutex>> locks;
            locks.reserve(kStripes);
            for (auto& st : kv_) locks.emplace_back(st.mu);
            for (auto& st : kv_)
                st.map.for_each([&](std::string_view, Ref<BlockEntry>& val) {
                    BlockEntry* e = val.get();
                    if (e->shard == shard && e->committed && e->ref_count() == 1) {
                        held.push_back(val);
                        movable.push_back({e->ptr, e->size});
                    }
                });
        }
        if (movable.empty()) continue;
        auto moves = shard->plan_compaction(movable);
        if (moves.empty()) continue;

        std::map<size_t, Shard::CopyJob> by_size;  // one job per page size
        for (auto& m : moves) {
            auto& j = by_size[m.size];
            j.bytes_per_block = m.size;
            j.src.push_back(reinterpret_cast<uint64_t>(m.old_ptr));
            j.dst.push_back(reinterpret_cast<uint64_t>(m.new_ptr));
        }
        bool copies_ok = true;
        for (auto& [sz, j2] : by_size) {
            std::promise<bool> cp;
            auto cf = cp.get_future();
            Shard::CopyJob jj = std::move(j2);
            jj.done = [&cp](bool ok) { cp.set_value(ok); };
            if (!shard->submit_copy(std::move(jj))) {
                cp.set_value(false);
            }
            if (!cf.get()) copies_ok = false;
        }

        std::map<void*, Shard::Move*> by_old;
        for (auto& m : moves) by_old[m.old_ptr] = &m;
        {
            std::vector<std::unique_lock<std::shared_mutex>> locks;
            locks.reserve(kStripes);
            for (auto& st : kv_) locks.emplace_back(st.mu);
            for (auto& st : kv_)
                st.map.for_each([&](std::string_view, Ref<BlockEntry>& val) {
                    BlockEntry* e = val.get();
                    auto it = by_old.find(e->ptr);
                    if (it == by_old.end() || e->shard != shard) return;
                    // map ref + our `held` ref = 2; more means an in-flight
                    // read captured the old pointer during the copy.
                    if (!copies_ok || e->ref_count() > 2) return;
                    Shard::Move* m = it->second;
                    e->ptr = m->new_ptr;
                    e->pool_idx = m->pool_idx;
                    shard->deallocate(m->old_ptr, m->size, m->pool_idx);
                    m->old_ptr = nullptr;  // applied
                    moved++;
                    bytes += m->size;
                });
        }
        for (auto& m : moves)
            if (m.old_ptr != nullptr)  // unapplied: release the planned slot
                shard->deallocate(m.new_ptr, m.size, m.pool_idx);
        // held refs drop here; entries erased during the copy free their
        // (old) blocks now.
    }
    return {moved, bytes};
}

size_t Server::kvmap_len() {
    size_t n = 0;
    for (auto& st : kv_) {
        std::shared_lock<std::shared_mutex> lk(st.mu);
        n += st.map.size();
    }
    return n;
}

size_t Server::purge() {
    size_t n = 0;
    for (auto& st : kv_) {
        std::lock_guard<std::shared_mutex> lk(st.mu);
        n += st.map.size();
        st.map.clear();
    }
    return n;
}

// ---- snapshot / restore ----------------------------------------------------
namespace {
constexpr uint64_t kSnapMagic = 0x53494653504e3150ull;  // "SIFSPN1P"
struct SnapEntryHdr {
    uint32_t key_len;
    uint32_t fp8;
    uint64_t size;  // stored bytes
    float scale;
    uint32_t _pad;
};
}  // namespace

bool Server::snapshot(const std::string& path, std::pair<size_t, size_t>* out) {
    *out = {0, 0};
    // Pin a consistent set of committed entries (refs keep blocks alive and
    // make compaction skip them — it requires ref_count()==1), then stream
    // them out without holding any lock.
    std::vector<std::pair<std::string, Ref<BlockEntry>>> pinned;
    for (auto& st : kv_) {
        std::shared_lock<std::shared_mutex> lk(st.mu);
        st.map.for_each([&](std::string_view key, Ref<BlockEntry>& val) {
            if (val->committed && !expired(val.get()))
                pinned.emplace_back(std::string(key), val);
        });
    }
    FILE* f = fopen(path.c_str(), "wb");
    if (!f) {
        ERROR("snapshot: cannot open %s", path.c_str());
        return false;
    }
    uint64_t magic = kSnapMagic, count = pinned.size();
    bool ok = fwrite(&magic, 8, 1, f) == 1 && fwrite(&count, 8, 1, f) == 1;
    std::vector<uint8_t> buf;
    for (auto& [key, ref] : pinned) {
        if (!ok) break;
        BlockEntry* e = ref.get();
        SnapEntryHdr h{static_cast<uint32_t>(key.size()), e->fp8 ? 1u : 0u, e->size,
                       e->scale, 0};
        buf.resize(e->size);
        if (e->shard->on_gpu()) {
            if (!gpu::memcpy_d2h(buf.data(), e->ptr, e->size)) {
                ok = false;
                break;
            }
        } else {
            memcpy(buf.data(), e->ptr, e->size);
        }
        ok = fwrite(&h, sizeof(h), 1, f) == 1 && fwrite(key.data(), 1, key.size(), f) == key.size() &&
             fwrite(buf.data(), 1, e->size, f) == e->size;
        out->first++;
        out->second += e->size;
    }
    fclose(f);
    if (!ok) ERROR("snapshot: write failed at entry %zu", out->first);
    return ok;
}

bool Server::restore(const std::string& path, std::pair<size_t, size_t>* out) {
    *out = {0, 0};
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) {
        ERROR("restore: cannot open %s", path.c_str());
        return false;
    }
    uint64_t magic = 0, count = 0;
    if (fread(&magic, 8, 1, f) != 1 || magic != kSnapMagic || fread(&count, 8, 1, f) != 1) {
        ERROR("restore: bad snapshot header");
        fclose(f);
        return false;
    }
    std::vector<uint8_t> buf;
    std::string key;
    bool ok = true;
    for (uint64_t i = 0; i < count && ok; i++) {
        SnapEntryHdr h{};
        if (fread(&h, sizeof(h), 1, f) != 1 || h.key_len > 4096 || h.size > (1u << 30)) {
            ok = false;
            break;
        }
        key.resize(h.key_len);
        buf.resize(h.size);
        if (fread(key.data(), 1, h.key_len, f) != h.key_len ||
            fread(buf.data(), 1, h.size, f) != h.size) {
            ok = false;
            break;
        }
        Shard* shard = shard_least_used();
        void* ptr = nullptr;
        int pool_idx = -1;
        if (!shard->allocate(h.size, 1, [&](void* p, int idx) {
                ptr = p;
                pool_idx = idx;
            })) {
            WARN("restore: pool full after %zu entries", out->first);
            break;  // partial restore is still useful
        }
        bool copied = shard->on_gpu() ? gpu::memcpy_h2d(ptr, buf.data(), h.size)
                                      : (memcpy(ptr, buf.data(), h.size), true);
        if (!copied) {
            shard->deallocate(ptr, h.size, pool_idx);
            ok = false;
            break;
        }
        auto* e = new BlockEntry();
        e->ptr = ptr;
        e->size = h.size;
        e->pool_idx = pool_idx;
        e->shard = shard;
        e->fp8 = h.fp8 != 0;
        e->scale = h.scale;
        e->born_sec = now_sec();
        e->committed = true;
        e->last_access.store(tick(), std::memory_order_relaxed);
        Ref<BlockEntry> ref(e);
        uint64_t hh = KvMap::hash_of(key);
        auto& st = kv_[stripe_of(hh)];
        bool ins = false;
        {
            std::lock_guard<std::shared_mutex> lk(st.mu);
            st.map.emplace_hashed(key, hh, ref, &ins);
        }
        if (ins) {
            out->first++;
            out->second += h.size;
        }
        // !ins: live key wins over the snapshot; `ref` frees the block.
    }
    fclose(f);
    return ok;
}

std::string Server::stats_json() {
    char buf[1024];
    size_t used = 0, total = 0, frag = 0;
    for (auto& s : shards_) {
        used += s->used_blocks();
        total += s->total_blocks();
        frag = std::max(frag, s->largest_free_run_bytes());
    }
    snprintf(buf, sizeof(buf),
             "{\"kv_len\": %zu, \"shards\": %zu, \"used_blocks\": %zu, \"total_blocks\": %zu, "
             "\"writes\": %llu, \"reads\": %llu, \"puts\": %llu, \"gets\": %llu, "
             "\"bytes_in\": %llu, \"bytes_out\": %llu, \"largest_free_run_bytes\": %zu, "
             "\"op_us\": {",
             kvmap_len(), shards_.size(), used, total,
             (unsigned long long)n_writes_.load(), (unsigned long long)n_reads_.load(),
             (unsigned long long)n_put_.load(), (unsigned long long)n_get_.load(),
             (unsigned long long)bytes_in_.load(), (unsigned long long)bytes_out_.load(), frag);
    std::string out(buf);
    bool first = true;
    for (int i = 0; i < 128; i++) {
        uint64_t cnt = op_stats_[i].count.load();
        if (!cnt) continue;
        char entry[160];
        snprintf(entry, sizeof(entry),
                 "%s\"%s\": {\"count\": %llu, \"avg_us\": %.1f, \"max_us\": %llu}",
                 first ? "" : ", ", op_name(static_cast<char>(i)).c_str(),
                 (unsigned long long)cnt,
                 static_cast<double>(op_stats_[i].total_us.load()) / cnt,
                 (unsigned long long)op_stats_[i].max_us.load());
        out += entry;
        first = false;
    }
    out += "}}";
    return out;
}

}  // namespace ifs
