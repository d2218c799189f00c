#include "client_verbs.h"

#include <poll.h>

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <mutex>
#include <thread>
#include <unordered_set>

#include "../core/log.h"
#include "../core/utils.h"
#include "../fabric/verbs_fabric.h"
#include "../fabric/wr_flow.h"
#include "../gpu/gpu.h"
#include "client.h"

namespace ifs {

namespace {
// One verbs driver per client process (device context + PD + MR table).
vf::Driver& client_driver() {
    static vf::Driver drv;
    return drv;
}
std::mutex g_drv_mu;
}  // namespace

struct VerbsClient::Impl {
    std::unique_ptr<vf::Endpoint> ep;
    std::unique_ptr<WrFlow> flow;
    std::thread poller;
    std::atomic<bool> stop{false};
    int wake_pipe[2] = {-1, -1};

    std::mutex mu;
    std::condition_variable cv;
    std::deque<int> free_send;
    std::vector<uint8_t> last_msg;  // latest SEND-channel response payload
    bool msg_ready = false;
    std::unordered_set<uint32_t> done_cookies;
    int imm_count = 0;
    uint32_t last_imm = 0;  // 0 = success; else server status (e.g. 404)
    uint32_t next_cookie = 1;
    bool error = false;

    std::mutex op_mu;  // serializes allocate/write/read ops

    ~Impl() {
        stop.store(true, std::memory_order_release);
        if (wake_pipe[1] >= 0) {
            char b = 1;
            ssize_t r = ::write(wake_pipe[1], &b, 1);
            (void)r;
        }
        if (poller.joinable()) poller.join();
        if (wake_pipe[0] >= 0) ::close(wake_pipe[0]);
        if (wake_pipe[1] >= 0) ::close(wake_pipe[1]);
    }

    void on_event(vf::Ev ev, uint64_t id, uint32_t imm, uint32_t len) {
        std::lock_guard<std::mutex> lk(mu);
        switch (ev) {
            case vf::Ev::kSendDone:
                free_send.push_back(static_cast<int>(id));
                break;
            case vf::Ev::kRecvMsg:
                last_msg.assign(ep->recv_buf(static_cast<int>(id)),
                                ep->recv_buf(static_cast<int>(id)) + len);
                msg_ready = true;
                ep->post_recv_buf(static_cast<int>(id));
                break;
            case vf::Ev::kRecvImm:
                imm_count++;
                last_imm = imm;  // nonzero = server-reported read failure
                break;
            case vf::Ev::kWriteDone: {
                uint32_t cookie = static_cast<uint32_t>(vf::Endpoint::write_cookie(id));
                flow->on_chain_complete(
                    static_cast<size_t>(vf::Endpoint::write_chain_len(id)));
                if (cookie) done_cookies.insert(cookie);
                break;
            }
            case vf::Ev::kError:
                error = true;
                break;
        }
        cv.notify_all();
    }

    void poll_loop() {
        struct pollfd fds[2];
        fds[0].fd = ep->comp_fd();
        fds[0].events = POLLIN;
        fds[1].fd = wake_pipe[0];
        fds[1].events = POLLIN;
        for (;;) {
            if (stop.load(std::memory_order_acquire)) return;
            int r = ::poll(fds, 2, 1000);
            if (r < 0 && errno != EINTR) return;
            if (fds[0].revents & POLLIN) {
                ep->drain([this](vf::Ev ev, uint64_t id, uint32_t imm, uint32_t len) {
                    on_event(ev, id, imm, len);
                });
            }
        }
    }

    template <typename Pred>
    bool wait_for(Pred pred, int seconds = 10) {
        std::unique_lock<std::mutex> lk(mu);
        return cv.wait_for(lk, std::chrono::seconds(seconds),
                           [&] { return error || pred(); }) &&
               !error;
    }

    int take_send_buf() {
        std::unique_lock<std::mutex> lk(mu);
        cv.wait_for(lk, std::chrono::seconds(10), [&] { return !free_send.empty(); });
        if (free_send.empty()) return -1;
        int i = free_send.front();
        free_send.pop_front();
        return i;
    }
};

VerbsClient::VerbsClient() : impl_(new Impl()) {}
VerbsClient::~VerbsClient() = default;

std::unique_ptr<VerbsClient> VerbsClient::establish(int fd, const ClientConfigC& cfg,
                                                    bool* attempted) {
    *attempted = false;
    if (!vf::compiled_in() || cfg.link_type == "TCP") {
        DEBUG("verbs skipped (compiled_in=%d link=%s)", vf::compiled_in() ? 1 : 0,
              cfg.link_type.c_str());
        return nullptr;
    }
    vf::Options o;
    o.dev_name = cfg.dev_name;
    o.ib_port = cfg.ib_port;
    o.roce = cfg.link_type != "IB";
    {
        std::lock_guard<std::mutex> lk(g_drv_mu);
        if (!client_driver().ready()) {
            if (!vf::device_available(o)) {
                DEBUG("verbs skipped: no usable RDMA device");
                return nullptr;
            }
            std::string err;
            if (!client_driver().init(o, &err)) {
                WARN("client verbs driver init failed: %s", err.c_str());
                return nullptr;
            }
        }
    }
    std::unique_ptr<VerbsClient> vc(new VerbsClient());
    auto& im = *vc->impl_;
    im.ep = std::make_unique<vf::Endpoint>(client_driver());
    vf::ConnInfo local{}, remote{};
    std::string err;
    if (!im.ep->init(&local, &err)) {
        WARN("client verbs endpoint init failed: %s", err.c_str());
        return nullptr;
    }

    // OP_RDMA_EXCHANGE with our ConnInfo; the server replies VRBS+ConnInfo
    // when it can serve verbs, TCPF when it cannot.
    *attempted = true;
    Header h{kMagic, OP_RDMA_EXCHANGE, static_cast<uint32_t>(sizeof(local))};
    if (!send_exact(fd, &h, sizeof(h)) || !send_exact(fd, &local, sizeof(local)))
        return nullptr;
    int code = 0;
    if (!recv_exact(fd, &code, 4) || code != FINISH) return nullptr;
    uint32_t plen = 0;
    if (!recv_exact(fd, &plen, 4)) return nullptr;
    std::vector<uint8_t> payload(plen);
    if (plen && !recv_exact(fd, payload.data(), plen)) return nullptr;
    if (plen < 4 + sizeof(remote) || memcmp(payload.data(), "VRBS", 4) != 0) {
        DEBUG("server declined verbs; using the TCP fabric");
        return nullptr;
    }
    memcpy(&remote, payload.data() + 4, sizeof(remote));
    if (!im.ep->connect(remote, &err)) {
        WARN("client verbs connect failed: %s", err.c_str());
        return nullptr;
    }
    for (int i = 0; i < vf::kRecvBufs; i++) im.ep->post_recv_buf(i);
    for (int i = 0; i < vf::kSendBufs; i++) im.free_send.push_back(i);
    im.flow = std::make_unique<WrFlow>(
        [ep = im.ep.get()](const WrChain& ch) { return ep->post_write_chain(ch); });
    if (pipe(im.wake_pipe) != 0) return nullptr;
    im.poller = std::thread([&im] { im.poll_loop(); });
    INFO("client verbs fabric established (qpn=%u)", local.qpn);
    return vc;
}

bool VerbsClient::register_mr(void* addr, size_t len, bool device_mem) {
    return client_driver().reg_region(addr, len, device_mem, nullptr);
}

std::vector<std::pair<uint32_t, uint64_t>> VerbsClient::allocate(
    const std::vector<std::string>& keys, int block_size) {
    auto& im = *impl_;
    std::lock_guard<std::mutex> op(im.op_mu);
    RemoteMetaMsg msg;
    msg.keys = keys;
    msg.block_size = block_size;
    msg.op = OP_RDMA_ALLOCATE;
    auto body = build_remote_meta(msg);
    if (body.size() > vf::kMsgBufSize) return {};
    int sb = im.take_send_buf();
    if (sb < 0) return {};
    {
        std::lock_guard<std::mutex> lk(im.mu);
        im.msg_ready = false;
    }
    memcpy(im.ep->send_buf(sb), body.data(), body.size());
    if (!im.ep->post_send_msg(sb, body.size())) return {};
    if (!im.wait_for([&] { return im.msg_ready; }, 5)) {
        WARN("verbs allocate timed out");
        return {};
    }
    std::vector<RemoteBlockWire> wire;
    {
        std::lock_guard<std::mutex> lk(im.mu);
        if (!parse_allocate_response(im.last_msg.data(), im.last_msg.size(), &wire)) return {};
    }
    std::vector<std::pair<uint32_t, uint64_t>> out;
    out.reserve(wire.size());
    for (auto& w : wire) out.push_back({w.rkey, w.remote_addr});
    return out;
}

int VerbsClient::write_blocks(const uint64_t* offsets, size_t n_offsets, int block_size,
                              const std::pair<uint32_t, uint64_t>* blocks, size_t n_blocks,
                              uintptr_t base_ptr) {
    if (n_offsets != n_blocks) return -1;
    auto& im = *impl_;
    std::lock_guard<std::mutex> op(im.op_mu);
    vf::MrInfo mr;
    if (!client_driver().lookup_region(reinterpret_cast<void*>(base_ptr), &mr)) {
        ERROR("verbs write: source region not registered (call register_mr)");
        return -1;
    }
    std::vector<WrDesc> wrs;
    std::vector<uint64_t> addrs;
    wrs.reserve(n_blocks);
    for (size_t i = 0; i < n_blocks; i++) {
        if (is_fake_remote_block(blocks[i].first, blocks[i].second)) continue;  // dup key
        wrs.push_back({base_ptr + offsets[i], blocks[i].second,
                       static_cast<uint32_t>(block_size), mr.lkey, blocks[i].first});
        addrs.push_back(blocks[i].second);
    }
    if (wrs.empty()) return 0;
    uint32_t cookie;
    {
        std::lock_guard<std::mutex> lk(im.mu);
        cookie = im.next_cookie++;
        if (!im.next_cookie) im.next_cookie = 1;
    }
    {
        std::lock_guard<std::mutex> lk(im.mu);
        if (!im.flow->submit(std::move(wrs), /*with_imm=*/false, 0, cookie)) return -1;
    }
    if (!im.wait_for([&] { return im.done_cookies.count(cookie) > 0; })) {
        WARN("verbs write timed out");
        return -1;
    }
    {
        std::lock_guard<std::mutex> lk(im.mu);
        im.done_cookies.erase(cookie);
    }
    // Commit message so readers see the blocks (server does not ACK on the
    // verbs fabric — reference semantics, libinfinistore.cpp:363-396).
    RemoteMetaMsg cm;
    cm.op = OP_RDMA_WRITE_COMMIT;
    cm.block_size = block_size;
    cm.remote_addrs = std::move(addrs);
    auto body = build_remote_meta(cm);
    int sb = im.take_send_buf();
    if (sb < 0 || body.size() > vf::kMsgBufSize) return -1;
    memcpy(im.ep->send_buf(sb), body.data(), body.size());
    return im.ep->post_send_msg(sb, body.size()) ? 0 : -1;
}

int VerbsClient::read_blocks(const std::vector<std::pair<std::string, uint64_t>>& blocks,
                             int block_size, uintptr_t base_ptr) {
    auto& im = *impl_;
    std::lock_guard<std::mutex> op(im.op_mu);
    vf::MrInfo mr;
    if (!client_driver().lookup_region(reinterpret_cast<void*>(base_ptr), &mr)) {
        ERROR("verbs read: destination region not registered (call register_mr)");
        return -1;
    }
    int imm_before;
    {
        std::lock_guard<std::mutex> lk(im.mu);
        imm_before = im.imm_count;
    }
    im.ep->post_recv_bare(0);  // catches the server's WRITE_WITH_IMM
    RemoteMetaMsg msg;
    msg.block_size = block_size;
    msg.rkey = mr.rkey;
    msg.op = OP_RDMA_READ;
    for (auto& b : blocks) {
        msg.keys.push_back(b.first);
        msg.remote_addrs.push_back(base_ptr + b.second);
    }
    auto body = build_remote_meta(msg);
    int sb = im.take_send_buf();
    if (sb < 0 || body.size() > vf::kMsgBufSize) return -1;
    memcpy(im.ep->send_buf(sb), body.data(), body.size());
    if (!im.ep->post_send_msg(sb, body.size())) return -1;
    if (!im.wait_for([&] { return im.imm_count > imm_before; })) {
        WARN("verbs read timed out");
        return -1;
    }
    {
        std::lock_guard<std::mutex> lk(im.mu);
        if (im.last_imm != 0) {
            int status = static_cast<int>(im.last_imm);
            WARN("verbs read failed: server status %d", status);
            im.last_imm = 0;
            return status == KEY_NOT_FOUND ? -KEY_NOT_FOUND : -1;
        }
    }
    return 0;
}

}  // namespace ifs
