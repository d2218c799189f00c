#include "client.h"

#include <arpa/inet.h>
#include <limits.h>
#include <netdb.h>
#include <sys/uio.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <cstring>

#include "../core/log.h"
#include "../core/utils.h"
#include "../gpu/gpu.h"
#include "client_verbs.h"

namespace ifs {

ClientConn::ClientConn() = default;
ClientConn::~ClientConn() { close_conn(); }

void ClientConn::close_conn() {
    {
        std::lock_guard<std::mutex> lk(q_mu_);
        worker_stop_ = true;
    }
    q_cv_.notify_all();
    if (worker_.joinable()) worker_.join();
    worker_started_ = false;
    verbs_.reset();
    if (shm_active_) {
        shm_active_ = false;
        shm_.unmap();  // server side is torn down by the socket close below
    }
    if (fd_ >= 0) {
        ::close(fd_);
        fd_ = -1;
    }
    connected_ = false;
    rdma_connected_ = false;
}

int ClientConn::init_connection(const ClientConfigC& cfg) {
    set_log_level(cfg.log_level.c_str());
    if (connected_) return -1;
    // Same-host fast path: the server exposes a Unix-domain socket next to
    // its TCP port (lower latency than TCP loopback; same wire protocol).
    if (cfg.host_addr == "127.0.0.1" || cfg.host_addr == "localhost") {
        std::string path = "/tmp/infinistore-amd-" + std::to_string(cfg.service_port) + ".sock";
        struct sockaddr_un ua{};
        if (path.size() < sizeof(ua.sun_path)) {
            int ufd = ::socket(AF_UNIX, SOCK_STREAM, 0);
            if (ufd >= 0) {
                ua.sun_family = AF_UNIX;
                strncpy(ua.sun_path, path.c_str(), sizeof(ua.sun_path) - 1);
                if (::connect(ufd, reinterpret_cast<struct sockaddr*>(&ua), sizeof(ua)) == 0) {
                    struct timeval tv{60, 0};
                    setsockopt(ufd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
                    int bufsz = 4 << 20;  // bulk fabric payloads
                    setsockopt(ufd, SOL_SOCKET, SO_SNDBUF, &bufsz, sizeof(bufsz));
                    setsockopt(ufd, SOL_SOCKET, SO_RCVBUF, &bufsz, sizeof(bufsz));
                    fd_ = ufd;
                    connected_ = true;
                    DEBUG("connected via UDS %s", path.c_str());
                    if (cfg.connection_type == "LOCAL_GPU" && !getenv("IFS_NO_SHM"))
                        try_shm_setup(cfg.service_port);
                    return 0;
                }
                ::close(ufd);
            }
        }
    }
    struct addrinfo hints{}, *res = nullptr;
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    char port[16];
    snprintf(port, sizeof(port), "%d", cfg.service_port);
    if (getaddrinfo(cfg.host_addr.c_str(), port, &hints, &res) != 0 || !res) {
        ERROR("getaddrinfo(%s) failed", cfg.host_addr.c_str());
        return -1;
    }
    int fd = ::socket(res->ai_family, res->ai_socktype, res->ai_protocol);
    if (fd < 0) {
        freeaddrinfo(res);
        return -1;
    }
    if (::connect(fd, res->ai_addr, res->ai_addrlen) != 0) {
        ERROR("connect %s:%d failed: %s", cfg.host_addr.c_str(), cfg.service_port,
              strerror(errno));
        ::close(fd);
        freeaddrinfo(res);
        return -1;
    }
    freeaddrinfo(res);
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    // Failure detection: a dead/stuck server surfaces as an op error after
    // 60 s instead of hanging the caller forever (the reference's client
    // blocks indefinitely on its sockets).
    struct timeval tv{60, 0};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    int bufsz = 4 << 20;  // bulk fabric payloads
    setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &bufsz, sizeof(bufsz));
    setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &bufsz, sizeof(bufsz));
    fd_ = fd;
    connected_ = true;
    if ((cfg.host_addr == "127.0.0.1" || cfg.host_addr == "localhost") &&
        cfg.connection_type == "LOCAL_GPU" && !getenv("IFS_NO_SHM"))
        try_shm_setup(cfg.service_port);
    return 0;
}

// ---------------------------------------------------------------------------
// shared-memory ring transport (same-host fast path for the packed ops)
// ---------------------------------------------------------------------------
bool ClientConn::try_shm_setup(int port) {
    static std::atomic<uint32_t> counter{0};
    std::string name = "/ifs-" + std::to_string(getpid()) + "-" + std::to_string(port) + "-" +
                       std::to_string(counter.fetch_add(1));
    if (!shmring::create_segment(name, 1u << 20, 64u << 10, &shm_)) return false;
    bool ok = false;
    if (send_req(OP_SHM_SETUP, reinterpret_cast<const uint8_t*>(name.data()), name.size())) {
        int code = 0;
        if (recv_status(&code) && code == FINISH) ok = true;
    }
    shm_unlink(name.c_str());  // both sides hold mappings; drop the name
    if (!ok) {
        shm_.unmap();
        DEBUG("shm ring setup declined; staying on the socket");
        return false;
    }
    shm_active_ = true;
    DEBUG("shm ring transport active");
    return true;
}

// Non-blocking: collect any responses already in the ring. Responses whose
// seq nobody is waiting for are async-write errors; remember the first.
void ClientConn::shm_drain_responses() {
    for (;;) {
        uint32_t len = 0;
        uint64_t skip = 0;
        const uint8_t* rec = shm_.resp->peek(&len, &skip);
        if (!rec) return;
        if (len >= sizeof(shmring::RespRec)) {
            shmring::RespRec r;
            memcpy(&r, rec, sizeof(r));
            if (shm_results_.size() < 4096) shm_results_[r.h.seq] = r.status;
            if (r.status != 0 && r.status != TASK_ACCEPTED && r.status != FINISH &&
                shm_async_err_ == 0)
                shm_async_err_ = r.status;
        }
        shm_.resp->consume(skip);
    }
}

int ClientConn::shm_wait(uint64_t seq) {
    auto it0 = shm_results_.find(seq);
    if (it0 != shm_results_.end()) {  // already drained while waiting elsewhere
        int st = it0->second;
        shm_results_.erase(it0);
        return st;
    }
    auto t0 = std::chrono::steady_clock::now();
    auto finish = [&](int st) {  // fold this wait into the spin-budget EWMA
        double us = std::chrono::duration<double, std::micro>(
                        std::chrono::steady_clock::now() - t0)
                        .count();
        shm_ewma_us_ = 0.75 * shm_ewma_us_ + 0.25 * std::min(us, 5000.0);
        return st;
    };
    // Adaptive spin: spin through ~2x the typical wait (usleep granularity
    // is ~50 µs and would dominate the 20-300 µs fast path), then back off
    // to sleeps so ms-scale waiters (64-client saturation) don't burn a
    // core each. Clock checked every 256 pauses.
    // ms-scale typical waits: spinning the first 400 µs is pure waste —
    // sleep almost immediately (the 50 µs usleep granularity is noise
    // against the wait itself).
    const double spin_deadline_us =
        shm_ewma_us_ > 800.0 ? 60.0 : std::min(400.0, std::max(30.0, shm_ewma_us_ * 2.0));
    int spins = 0;
    bool spinning = true;
    for (;;) {
        uint32_t len = 0;
        uint64_t skip = 0;
        const uint8_t* rec = shm_.resp->peek(&len, &skip);
        if (rec) {
            if (len == 0) {  // wrap marker
                shm_.resp->consume(skip);
                continue;
            }
            shmring::RespRec r{};
            memcpy(&r, rec, std::min(sizeof(r), size_t(len)));
            shm_.resp->consume(skip);
            if (r.h.seq == seq) return finish(r.status);
            // Out-of-order response: stash for a later wait (bounded — a
            // runaway map means tickets are being dropped by the caller).
            if (shm_results_.size() < 4096) shm_results_[r.h.seq] = r.status;
            if (r.status != 0 && r.status != TASK_ACCEPTED && r.status != FINISH &&
                shm_async_err_ == 0)
                shm_async_err_ = r.status;
            continue;
        }
        if (spinning) {
#if defined(__x86_64__)
            __builtin_ia32_pause();
#endif
            if ((++spins & 255) == 0) {
                auto us = std::chrono::duration<double, std::micro>(
                              std::chrono::steady_clock::now() - t0)
                              .count();
                if (us > spin_deadline_us) spinning = false;
            }
        } else {
            usleep(50);
            if (std::chrono::steady_clock::now() - t0 > std::chrono::seconds(60)) {
                ERROR("shm ring response timeout (seq %llu)",
                      static_cast<unsigned long long>(seq));
                return kShmErr;
            }
        }
    }
}

int ClientConn::shm_request(char op, const uint8_t* body, size_t n, bool want_resp,
                            uint64_t* out_ticket) {
    uint32_t need = shmring::rec_len(n);
    if (need > shm_.req->cap / 2) return kShmNoFit;
    shm_drain_responses();
    uint64_t seq = ++shm_seq_;
    uint64_t adv = 0;
    uint8_t* dst = shm_.req->claim(need, &adv);
    auto t0 = std::chrono::steady_clock::now();
    while (!dst) {  // ring full: the server is behind; wait for space
        shm_drain_responses();
#if defined(__x86_64__)
        __builtin_ia32_pause();
#endif
        if (std::chrono::steady_clock::now() - t0 > std::chrono::seconds(60)) return kShmErr;
        dst = shm_.req->claim(need, &adv);
    }
    shmring::RecHdr h{};
    h.len = need;
    h.op = static_cast<uint8_t>(op);
    h.body_len = static_cast<uint32_t>(n);
    h.t_push_us = shmring::mono_us();
    h.seq = seq;
    memcpy(dst, &h, sizeof(h));
    if (n) memcpy(dst + sizeof(h), body, n);
    shm_.req->publish(adv);
    if (!want_resp) {
        shm_unacked_++;
        return 0;
    }
    if (out_ticket) {  // ticketed: the caller waits later (wait_local_ticket)
        *out_ticket = seq;
        shm_unacked_++;
        return 0;
    }
    int code = shm_wait(seq);
    shm_unacked_ = 0;  // a sync-response op drains the ring ordering-wise
    return code;
}

int ClientConn::wait_local_ticket(uint64_t ticket) {
    if (ticket == 0) return 0;
    std::lock_guard<std::mutex> lk(io_mu_);
    if (!shm_active_) return -1;
    int code = shm_wait(ticket);
    if (code == kShmErr) return -1;
    if (code != TASK_ACCEPTED && code != FINISH && code != 0) return code < 0 ? code : -code;
    return 0;
}

int ClientConn::shm_ring_sync() { return shm_request(OP_SYNC, nullptr, 0, /*want_resp=*/true); }

int ClientConn::setup_rdma(const ClientConfigC& cfg) {
    if (!connected_) return -1;
    std::lock_guard<std::mutex> lk(io_mu_);
    // Prefer the verbs fabric when rdma-core + a NIC are available on both
    // ends; otherwise (or on any handshake failure) the TCP data fabric
    // serves the same API.
    bool attempted = false;
    verbs_ = VerbsClient::establish(fd_, cfg, &attempted);
    if (!verbs_ && !attempted) {
        if (!send_req(OP_RDMA_EXCHANGE, nullptr, 0)) return -1;
        int code = 0;
        if (!recv_status(&code) || code != FINISH) return -1;
        std::vector<uint8_t> tag;
        if (!recv_payload(&tag) || tag.size() < 4) return -1;
        if (memcmp(tag.data(), "TCPF", 4) != 0) {
            ERROR("unknown fabric tag");
            return -1;
        }
    }
    rdma_connected_ = true;
    // Spawn the async worker lazily on first connect.
    if (!worker_started_) {
        worker_stop_ = false;
        worker_ = std::thread([this] { worker_main(); });
        worker_started_ = true;
    }
    return 0;
}

// ---------------------------------------------------------------------------
// framing helpers (io_mu_ must be held)
// ---------------------------------------------------------------------------
namespace {
// Gathered send: avoids assembling multi-MB bodies just to copy them into
// the socket (the TCP fabric's put path sends tensor pages directly).
bool recv_iov(int fd, struct iovec* iov, int cnt) {
    while (cnt > 0) {
        ssize_t r = ::readv(fd, iov, std::min(cnt, IOV_MAX));
        if (r <= 0) {
            if (r < 0 && errno == EINTR) continue;
            return false;
        }
        size_t left = static_cast<size_t>(r);
        while (cnt > 0 && left >= iov->iov_len) {
            left -= iov->iov_len;
            iov++;
            cnt--;
        }
        if (cnt > 0 && left) {
            iov->iov_base = static_cast<uint8_t*>(iov->iov_base) + left;
            iov->iov_len -= left;
        }
    }
    return true;
}

bool send_iov(int fd, struct iovec* iov, int cnt) {
    while (cnt > 0) {
        ssize_t w = ::writev(fd, iov, std::min(cnt, IOV_MAX));
        if (w < 0) {
            if (errno == EINTR) continue;
            return false;
        }
        size_t left = static_cast<size_t>(w);
        while (cnt > 0 && left >= iov->iov_len) {
            left -= iov->iov_len;
            iov++;
            cnt--;
        }
        if (cnt > 0 && left) {
            iov->iov_base = static_cast<uint8_t*>(iov->iov_base) + left;
            iov->iov_len -= left;
        }
    }
    return true;
}
}  // namespace

bool ClientConn::send_req(char op, const uint8_t* body, size_t n) {
    // Ring requests and socket requests are handled by different server
    // threads; a ring sync round trip here restores the total order the
    // socket alone used to give (only needed when ring ops are unacked).
    if (shm_active_ && shm_unacked_ > 0) shm_ring_sync();
    Header h{kMagic, op, static_cast<uint32_t>(n)};
    if (!send_exact(fd_, &h, sizeof(h))) return false;
    if (n && !send_exact(fd_, body, n)) return false;
    return true;
}

bool ClientConn::recv_status(int* code) { return recv_exact(fd_, code, 4); }

bool ClientConn::recv_payload(std::vector<uint8_t>* out) {
    uint32_t len = 0;
    if (!recv_exact(fd_, &len, 4)) return false;
    out->resize(len);
    if (len && !recv_exact(fd_, out->data(), len)) return false;
    return true;
}

// ---------------------------------------------------------------------------
// local (IPC) path
// ---------------------------------------------------------------------------
int ClientConn::rw_local(char op, const std::vector<std::pair<std::string, uint64_t>>& blocks,
                         int block_size, uintptr_t ptr, int device_id) {
    std::string blob;
    std::vector<uint64_t> offs;
    offs.reserve(blocks.size());
    for (size_t i = 0; i < blocks.size(); i++) {
        if (i) blob.push_back('\0');
        blob.append(blocks[i].first);
        offs.push_back(blocks[i].second);
    }
    return rw_local_packed(op, blob.data(), blob.size(), offs.data(), blocks.size(), block_size,
                           ptr, device_id, false);
}

int ClientConn::rw_local_packed(char op, const char* keys_blob, size_t blob_len,
                                const uint64_t* offsets, size_t n, int block_size,
                                uintptr_t ptr, int device_id, bool sync_response,
                                uint64_t* out_ticket, uint32_t extra_flags) {
    if (out_ticket) *out_ticket = 0;  // 0 = completed synchronously
    if (!connected_) return -1;
    if (!gpu::available()) {
        ERROR("local path requires a GPU");
        return -1;
    }
    gpu::IpcHandle handle;
    uint64_t base_offset = 0;
    uint64_t alloc_size = 0;
    {
        std::lock_guard<std::mutex> lk(ipc_mu_);
        auto it = ipc_export_cache_.find(ptr);
        if (it != ipc_export_cache_.end()) {
            memcpy(handle.bytes, it->second.handle, gpu::kIpcHandleSize);
            base_offset = it->second.base_offset;
            alloc_size = it->second.alloc_size;
        } else {
            bool have = gpu::ipc_export(reinterpret_cast<void*>(ptr), &handle,
                                        &base_offset, &alloc_size);
            if (!have) memset(handle.bytes, 0, gpu::kIpcHandleSize);
            if (ipc_export_cache_.size() > 4096) ipc_export_cache_.clear();
            IpcExport ent;
            memcpy(ent.handle, handle.bytes, gpu::kIpcHandleSize);
            ent.base_offset = base_offset;
            ent.alloc_size = alloc_size;
            ent.have_handle = have;
            ipc_export_cache_.emplace(ptr, ent);
        }
    }

    PackedLocalHdr h;
    h.device = device_id;
    h.pid = static_cast<int32_t>(getpid());
    h.base_ptr = ptr - base_offset;
    h.base_offset = base_offset;
    h.block_size = static_cast<uint32_t>(block_size);
    h.n_blocks = static_cast<uint32_t>(n);
    // Reads complete in one round trip (response deferred to completion);
    // writes default to async so uploads overlap compute (prefill pattern),
    // with an opt-in single-round-trip mode (write_pages(sync=True)).
    h.flags = ((op == 'R' || sync_response) ? kLocalFlagSyncResponse : 0) | extra_flags;
    // Containing allocation size in MB: the server refuses the IPC open for
    // allocations >= 2 GiB (hipIpcOpenMemHandle hangs importing them under
    // dmabuf IPC — scripts/ipc_size_probe.py); the same-process pid fast
    // path is exempt. 0 = unknown.
    h.rsvd = static_cast<uint32_t>((alloc_size + (1 << 20) - 1) >> 20);
    memcpy(h.ipc, handle.bytes, gpu::kIpcHandleSize);

    std::vector<uint8_t> body(sizeof(h) + n * 8 + blob_len);
    memcpy(body.data(), &h, sizeof(h));
    memcpy(body.data() + sizeof(h), offsets, n * 8);
    memcpy(body.data() + sizeof(h) + n * 8, keys_blob, blob_len);

    char wire_op = (op == 'W') ? OP_W_FAST : OP_R_FAST;
    static const bool dbg = getenv("IFS_CLIENT_DEBUG") != nullptr;
    auto t0 = std::chrono::steady_clock::now();
    std::lock_guard<std::mutex> lk(io_mu_);
    if (shm_active_) {
        bool want_resp = (h.flags & kLocalFlagSyncResponse) != 0;
        int code = shm_request(wire_op, body.data(), body.size(), want_resp,
                               want_resp ? out_ticket : nullptr);
        if (code != kShmNoFit) {
            if (out_ticket && *out_ticket) return 0;  // ticketed: wait later
            if (dbg && n > 64) {
                auto t2 = std::chrono::steady_clock::now();
                fprintf(stderr, "[cdbg] op=%c n=%zu shm total=%.0fus\n", op, n,
                        std::chrono::duration<double, std::micro>(t2 - t0).count());
            }
            if (want_resp && code != TASK_ACCEPTED && code != FINISH) {
                WARN("rw_local(shm) op=%c -> %d", op, code);
                return code < 0 ? code : -code;
            }
            if (op == 'W' && !sync_response) local_dirty_ = true;
            return 0;
        }
        // falls through to the socket (record too large for the ring);
        // send_req's ring sync keeps ordering.
    }
    if (!send_req(wire_op, body.data(), body.size())) return -1;
    auto t1 = std::chrono::steady_clock::now();
    int code = 0;
    if (!recv_status(&code)) return -1;
    if (dbg && n > 64) {
        auto t2 = std::chrono::steady_clock::now();
        fprintf(stderr, "[cdbg] op=%c n=%zu send=%.0fus wait=%.0fus\n", op, n,
                std::chrono::duration<double, std::micro>(t1 - t0).count(),
                std::chrono::duration<double, std::micro>(t2 - t1).count());
    }
    if (code != TASK_ACCEPTED && code != FINISH) {
        WARN("rw_local op=%c -> %d", op, code);
        return -code;
    }
    if (op == 'W' && !sync_response) local_dirty_ = true;  // sync RTT needed
    return 0;
}

int ClientConn::sync_local() {
    if (!connected_) return -1;
    std::lock_guard<std::mutex> lk(io_mu_);
    // Reads complete before their response (kLocalFlagSyncResponse), so the
    // sync round trip is only needed after writes.
    if (!local_dirty_ && shm_unacked_ == 0) return 0;
    if (shm_active_) {
        int code = shm_ring_sync();
        if (code == kShmErr) return -1;
        if (code != 0) return code < 0 ? code : -code;
        local_dirty_ = false;
        if (shm_async_err_) {  // surface an earlier async ring write failure
            int e = shm_async_err_;
            shm_async_err_ = 0;
            return -e;
        }
        return 0;
    }
    if (!send_req(OP_SYNC, nullptr, 0)) return -1;
    // Reference framing: FINISH + int remain (infinistore.cpp:1070-1075,
    // libinfinistore.cpp:632-657).
    int code = -1, remain = -1;
    if (!recv_status(&code)) return -1;
    if (code != FINISH) return code > 0 ? -code : code;
    if (!recv_exact(fd_, &remain, 4)) return -1;
    if (remain == 0) local_dirty_ = false;
    return remain;
}

// ---------------------------------------------------------------------------
// RDMA-semantics path over the TCP fabric
// ---------------------------------------------------------------------------
int ClientConn::register_mr(uintptr_t ptr, size_t size) {
    bool dev = false;
    if (gpu::available()) {
        // classify host vs device like the reference's cudaPointerGetAttributes
        // check (libinfinistore.cpp:1168)
        dev = gpu::is_device_pointer(reinterpret_cast<const void*>(ptr));
    }
    if (verbs_ && !verbs_->register_mr(reinterpret_cast<void*>(ptr), size, dev)) return -1;
    std::lock_guard<std::mutex> lk(region_mu_);
    regions_.push_back({ptr, size, dev});
    return 1;
}

bool ClientConn::is_device_ptr(uintptr_t ptr) {
    {
        std::lock_guard<std::mutex> lk(region_mu_);
        for (auto& r : regions_) {
            if (ptr >= r.ptr && ptr < r.ptr + r.size) return r.device;
        }
    }
    if (gpu::available()) return gpu::is_device_pointer(reinterpret_cast<const void*>(ptr));
    return false;
}

std::vector<RemoteBlockOut> ClientConn::do_allocate(const std::vector<std::string>& keys,
                                                    int block_size) {
    if (verbs_) {
        auto res = verbs_->allocate(keys, block_size);
        std::vector<RemoteBlockOut> out;
        out.reserve(res.size());
        for (auto& r : res) out.push_back({r.first, r.second});
        return out;
    }
    RemoteMetaMsg msg;
    msg.keys = keys;
    msg.block_size = block_size;
    msg.op = OP_RDMA_ALLOCATE;
    auto body = build_remote_meta(msg);
    std::lock_guard<std::mutex> lk(io_mu_);
    if (!send_req(OP_RDMA_ALLOCATE, body.data(), body.size())) return {};
    int code = 0;
    if (!recv_status(&code) || code != FINISH) {
        WARN("allocate -> %d", code);
        return {};
    }
    std::vector<uint8_t> payload;
    if (!recv_payload(&payload)) return {};
    std::vector<RemoteBlockWire> wire;
    if (!parse_allocate_response(payload.data(), payload.size(), &wire)) return {};
    std::vector<RemoteBlockOut> out;
    out.reserve(wire.size());
    for (auto& w : wire) out.push_back({w.rkey, w.remote_addr});
    return out;
}

std::vector<RemoteBlockOut> ClientConn::allocate_rdma(const std::vector<std::string>& keys,
                                                      int block_size) {
    if (!rdma_connected_) return {};
    return do_allocate(keys, block_size);
}

int ClientConn::allocate_rdma_async(const std::vector<std::string>& keys, int block_size,
                                    std::function<void(std::vector<RemoteBlockOut>)> cb) {
    if (!rdma_connected_) return -1;
    inflight_.fetch_add(1);
    enqueue([this, keys, block_size, cb = std::move(cb)] {
        auto res = do_allocate(keys, block_size);
        cb(std::move(res));
    });
    return 0;
}

int ClientConn::do_w_rdma(const uint64_t* offsets, size_t n_offsets, int block_size,
                          const RemoteBlockOut* blocks, size_t n_blocks, uintptr_t base_ptr) {
    if (n_offsets != n_blocks) return -1;
    if (verbs_) {
        std::vector<std::pair<uint32_t, uint64_t>> blks;
        blks.reserve(n_blocks);
        for (size_t i = 0; i < n_blocks; i++)
            blks.push_back({blocks[i].rkey, blocks[i].remote_addr});
        return verbs_->write_blocks(offsets, n_offsets, block_size, blks.data(), blks.size(),
                                    base_ptr);
    }
    bool dev = is_device_ptr(base_ptr);
    size_t bs = static_cast<size_t>(block_size);

    // Collect non-FAKE blocks (duplicates are skipped — first write wins).
    std::vector<size_t> idx;
    idx.reserve(n_blocks);
    for (size_t i = 0; i < n_blocks; i++) {
        if (!is_fake_remote_block(blocks[i].rkey, blocks[i].remote_addr)) idx.push_back(i);
    }

    // Chunked pipelined puts: assemble [n][bs][addrs][payload] bodies of up
    // to ~8 MB, stream them all, then collect the acks.
    const size_t kChunkBytes = 8u << 20;
    size_t per = std::max<size_t>(1, kChunkBytes / (bs + 8));
    size_t n_chunks = 0;
    std::vector<uint64_t> all_addrs;
    {
        std::unique_lock<std::mutex> lk(io_mu_);
        std::vector<uint8_t> body;       // assembly fallback (device tensors)
        std::vector<uint8_t> prefix;     // header+addrs for the gathered path
        std::vector<struct iovec> iov;
        for (size_t start = 0; start < idx.size(); start += per) {
            size_t take = std::min(per, idx.size() - start);
            // Host tensors: gathered send straight from the tensor pages —
            // no multi-MB body assembly. (Device tensors still stage D2H;
            // iovec count is bounded by IOV_MAX.)
            bool gather = !dev && take + 2 <= static_cast<size_t>(IOV_MAX);
            if (gather) {
                prefix.resize(sizeof(Header) + 8 + take * 8);
                auto* h = reinterpret_cast<Header*>(prefix.data());
                *h = Header{kMagic, OP_TCP_PUT,
                            static_cast<uint32_t>(8 + take * 8 + take * bs)};
                uint32_t n32 = static_cast<uint32_t>(take), bs32 = static_cast<uint32_t>(bs);
                memcpy(prefix.data() + sizeof(Header), &n32, 4);
                memcpy(prefix.data() + sizeof(Header) + 4, &bs32, 4);
                uint64_t* addrs =
                    reinterpret_cast<uint64_t*>(prefix.data() + sizeof(Header) + 8);
                iov.clear();
                iov.push_back({prefix.data(), prefix.size()});
                for (size_t j = 0; j < take; j++) {
                    size_t i = idx[start + j];
                    addrs[j] = blocks[i].remote_addr;
                    all_addrs.push_back(blocks[i].remote_addr);
                    iov.push_back({reinterpret_cast<void*>(base_ptr + offsets[i]), bs});
                }
                if (!send_iov(fd_, iov.data(), static_cast<int>(iov.size()))) return -1;
            } else {
                body.resize(8 + take * 8 + take * bs);
                uint32_t n32 = static_cast<uint32_t>(take), bs32 = static_cast<uint32_t>(bs);
                memcpy(body.data(), &n32, 4);
                memcpy(body.data() + 4, &bs32, 4);
                uint64_t* addrs = reinterpret_cast<uint64_t*>(body.data() + 8);
                uint8_t* payload = body.data() + 8 + take * 8;
                for (size_t j = 0; j < take; j++) {
                    size_t i = idx[start + j];
                    addrs[j] = blocks[i].remote_addr;
                    all_addrs.push_back(blocks[i].remote_addr);
                    const void* src = reinterpret_cast<const void*>(base_ptr + offsets[i]);
                    if (dev) {
                        if (!gpu::memcpy_d2h(payload + j * bs, src, bs)) return -1;
                    } else {
                        memcpy(payload + j * bs, src, bs);
                    }
                }
                if (!send_req(OP_TCP_PUT, body.data(), body.size())) return -1;
            }
            n_chunks++;
        }
        for (size_t ci = 0; ci < n_chunks; ci++) {
            int code = 0;
            if (!recv_status(&code)) return -1;
            if (code != TASK_ACCEPTED && code != FINISH) return -code;
        }
        if (!all_addrs.empty()) {
            // Commit; the server ACKs after flipping committed=true, so a
            // successful return here means readers will see the data.
            RemoteMetaMsg cm;
            cm.op = OP_RDMA_WRITE_COMMIT;
            cm.block_size = block_size;
            cm.remote_addrs = std::move(all_addrs);
            auto cbody = build_remote_meta(cm);
            if (!send_req(OP_RDMA_WRITE_COMMIT, cbody.data(), cbody.size())) return -1;
            int code = 0;
            if (!recv_status(&code) || code != FINISH) return -1;
        }
    }
    return 0;
}

int ClientConn::w_rdma(const uint64_t* offsets, size_t n_offsets, int block_size,
                       const RemoteBlockOut* blocks, size_t n_blocks, uintptr_t base_ptr) {
    if (!rdma_connected_) return -1;
    return do_w_rdma(offsets, n_offsets, block_size, blocks, n_blocks, base_ptr);
}

int ClientConn::w_rdma_async(const uint64_t* offsets, size_t n_offsets, int block_size,
                             const RemoteBlockOut* blocks, size_t n_blocks, uintptr_t base_ptr,
                             std::function<void()> cb) {
    if (!rdma_connected_) return -1;
    std::vector<uint64_t> offs(offsets, offsets + n_offsets);
    std::vector<RemoteBlockOut> blks(blocks, blocks + n_blocks);
    inflight_.fetch_add(1);
    enqueue([this, offs = std::move(offs), block_size, blks = std::move(blks), base_ptr,
             cb = std::move(cb)] {
        do_w_rdma(offs.data(), offs.size(), block_size, blks.data(), blks.size(), base_ptr);
        cb();
    });
    return 0;
}

int ClientConn::do_r_rdma(const std::vector<std::pair<std::string, uint64_t>>& blocks,
                          int block_size, uintptr_t base_ptr) {
    if (verbs_) return verbs_->read_blocks(blocks, block_size, base_ptr);
    bool dev = is_device_ptr(base_ptr);
    size_t bs = static_cast<size_t>(block_size);
    RemoteMetaMsg msg;
    msg.block_size = block_size;
    msg.op = OP_RDMA_READ;
    for (auto& b : blocks) msg.keys.push_back(b.first);
    auto body = build_remote_meta(msg);

    std::vector<uint8_t> payload;
    {
        std::lock_guard<std::mutex> lk(io_mu_);
        if (!send_req(OP_TCP_GET, body.data(), body.size())) return -1;
        int code = 0;
        if (!recv_status(&code)) return -1;
        if (code != FINISH) {
            WARN("read -> %d", code);
            return -code;
        }
        uint32_t len = 0;
        if (!recv_exact(fd_, &len, 4)) return -1;
        // Host tensors: scatter the payload straight into the destination
        // pages with readv (the payload is the request's blocks in order) —
        // no intermediate buffer or per-block memcpy.
        if (!dev && len == blocks.size() * bs) {
            std::vector<struct iovec> iov;
            iov.reserve(blocks.size());
            for (auto& b : blocks)
                iov.push_back({reinterpret_cast<void*>(base_ptr + b.second), bs});
            if (!recv_iov(fd_, iov.data(), static_cast<int>(iov.size()))) return -1;
            return 0;
        }
        payload.resize(len);
        if (len && !recv_exact(fd_, payload.data(), len)) return -1;
    }
    if (payload.size() < blocks.size() * bs) return -1;
    for (size_t i = 0; i < blocks.size(); i++) {
        void* dst = reinterpret_cast<void*>(base_ptr + blocks[i].second);
        const void* src = payload.data() + i * bs;
        if (dev) {
            if (!gpu::memcpy_h2d(dst, src, bs)) return -1;
        } else {
            memcpy(dst, src, bs);
        }
    }
    return 0;
}

int ClientConn::r_rdma(const std::vector<std::pair<std::string, uint64_t>>& blocks, int block_size,
                       uintptr_t base_ptr) {
    if (!rdma_connected_) return -1;
    return do_r_rdma(blocks, block_size, base_ptr);
}

int ClientConn::r_rdma_async(const std::vector<std::pair<std::string, uint64_t>>& blocks,
                             int block_size, uintptr_t base_ptr, std::function<void()> cb) {
    if (!rdma_connected_) return -1;
    inflight_.fetch_add(1);
    enqueue([this, blocks, block_size, base_ptr, cb = std::move(cb)] {
        do_r_rdma(blocks, block_size, base_ptr);
        cb();
    });
    return 0;
}

int ClientConn::sync_rdma() {
    std::unique_lock<std::mutex> lk(q_mu_);
    bool ok = drain_cv_.wait_for(lk, std::chrono::seconds(10),
                                 [this] { return inflight_.load() == 0; });
    return ok ? 0 : -1;
}

// ---------------------------------------------------------------------------
// queries
// ---------------------------------------------------------------------------
int ClientConn::check_exist(const std::string& key) {
    if (!connected_) return -1;
    // Reference framing: body = raw key bytes (body_size IS the key length,
    // libinfinistore.cpp:659-671); response FINISH + int (0 exists, 1 not).
    const uint8_t* body = reinterpret_cast<const uint8_t*>(key.data());
    std::lock_guard<std::mutex> lk(io_mu_);
    // Query ops ride the shm ring too (the reply status carries the value);
    // prefix lookups sit on the decode critical path.
    if (shm_active_) {
        int code = shm_request(OP_CHECK_EXIST, body, key.size(), true);
        if (code != kShmNoFit) return code == kShmErr ? -1 : code;
    }
    if (!send_req(OP_CHECK_EXIST, body, key.size())) return -1;
    int code = -1, exist = -1;
    if (!recv_status(&code)) return -1;
    if (code != FINISH) return -1;
    if (!recv_exact(fd_, &exist, 4)) return -1;
    return exist;  // 0 exists, 1 not
}

int ClientConn::delete_keys(const std::vector<std::string>& keys) {
    if (!connected_) return -1;
    auto body = build_match_request(keys);
    std::lock_guard<std::mutex> lk(io_mu_);
    if (shm_active_) {
        int code = shm_request(OP_DELETE, body.data(), body.size(), true);
        if (code != kShmNoFit) return code == kShmErr ? -1 : code;
    }
    if (!send_req(OP_DELETE, body.data(), body.size())) return -1;
    int n = -1;
    if (!recv_status(&n)) return -1;
    return n;
}

std::string ClientConn::get_stats() {
    if (!connected_) return "{}";
    std::lock_guard<std::mutex> lk(io_mu_);
    if (!send_req(OP_STATS, nullptr, 0)) return "{}";
    int code = 0;
    if (!recv_status(&code) || code != FINISH) return "{}";
    std::vector<uint8_t> payload;
    if (!recv_payload(&payload)) return "{}";
    return std::string(payload.begin(), payload.end());
}

int ClientConn::get_match_last_index(const std::vector<std::string>& keys) {
    if (!connected_) return -1;
    auto body = build_match_request(keys);
    std::lock_guard<std::mutex> lk(io_mu_);
    if (shm_active_) {
        int code = shm_request(OP_GET_MATCH_LAST_IDX, body.data(), body.size(), true);
        if (code != kShmNoFit) return code == kShmErr ? -1 : code;
    }
    if (!send_req(OP_GET_MATCH_LAST_IDX, body.data(), body.size())) return -1;
    // Reference framing: FINISH + int index (infinistore.cpp:1092-1108).
    int code = -1, idx = -1;
    if (!recv_status(&code)) return -1;
    if (code != FINISH) return -1;
    if (!recv_exact(fd_, &idx, 4)) return -1;
    return idx;
}

// ---------------------------------------------------------------------------
// async worker
// ---------------------------------------------------------------------------
void ClientConn::enqueue(std::function<void()> fn) {
    {
        std::lock_guard<std::mutex> lk(q_mu_);
        q_.push_back(std::move(fn));
    }
    q_cv_.notify_one();
}

void ClientConn::worker_main() {
    for (;;) {
        std::function<void()> fn;
        {
            std::unique_lock<std::mutex> lk(q_mu_);
            q_cv_.wait(lk, [this] { return worker_stop_ || !q_.empty(); });
            if (worker_stop_ && q_.empty()) return;
            fn = std::move(q_.front());
            q_.pop_front();
        }
        fn();
        if (inflight_.fetch_sub(1) == 1) {
            std::lock_guard<std::mutex> lk(q_mu_);
            drain_cv_.notify_all();
        }
    }
}

}  // namespace ifs
