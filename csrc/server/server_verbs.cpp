// Server-side verbs data plane: one RC QP per connection, CQ completions
// pumped through uv_poll on the libuv loop (the reference's design,
// infinistore.cpp:872-1052 / cq_poll_handle :212-334), dispatching the same
// allocate/commit/read logic as the TCP fabric. Reads are server-initiated
// RDMA_WRITE chains into client memory, flow-controlled by WrFlow, the set's
// final WR a WRITE_WITH_IMM. Inert unless rdma-core + a NIC are present
// (vf::compiled_in/device probing); see verbs_fabric.h for caveats.
#include <cstring>
#include <deque>
#include <unordered_map>

#include "../core/log.h"
#include "server.h"

namespace ifs {

struct Server::VerbsPeer {
    Server* srv;
    Server::Conn* conn;
    vf::Endpoint ep;
    uv_poll_t poll{};
    bool polling = false;
    WrFlow flow;
    std::deque<int> free_send;
    // cookie -> block refs held until the read's write chains complete
    std::unordered_map<uint32_t, std::shared_ptr<std::vector<Ref<BlockEntry>>>> reads;
    uint32_t next_cookie = 1;

    VerbsPeer(Server* s, Server::Conn* c)
        : srv(s),
          conn(c),
          ep(*s->vdrv_),
          flow([this](const WrChain& ch) { return ep.post_write_chain(ch); }) {}

    void on_event(vf::Ev ev, uint64_t id, uint32_t imm, uint32_t len) {
        switch (ev) {
            case vf::Ev::kSendDone:
                free_send.push_back(static_cast<int>(id));
                break;
            case vf::Ev::kRecvMsg:
                handle_msg(static_cast<int>(id), len);
                ep.post_recv_buf(static_cast<int>(id));
                break;
            case vf::Ev::kWriteDone: {
                uint32_t cookie = static_cast<uint32_t>(vf::Endpoint::write_cookie(id));
                size_t n = static_cast<size_t>(vf::Endpoint::write_chain_len(id));
                flow.on_chain_complete(n);
                if (cookie) reads.erase(cookie);  // releases the block refs
                break;
            }
            case vf::Ev::kRecvImm:
                break;  // not used server-side (client writes are unsignaled here)
            case vf::Ev::kError:
                WARN("verbs peer CQE error (conn will be torn down on TCP close)");
                break;
        }
        (void)imm;
    }

    void handle_msg(int buf, uint32_t len) {
        RemoteMetaMsg msg;
        if (!parse_remote_meta(ep.recv_buf(buf), len, &msg)) {
            WARN("verbs: bad RemoteMetaRequest");
            return;
        }
        if (msg.op == OP_RDMA_ALLOCATE) {
            int status = FINISH;
            auto blocks = srv->allocate_blocks(conn, msg.keys,
                                               static_cast<size_t>(msg.block_size), &status);
            auto payload = build_allocate_response(blocks);
            if (free_send.empty() || payload.size() > vf::kMsgBufSize) {
                ERROR("verbs: no send buffer for allocate response");
                return;
            }
            int sb = free_send.front();
            free_send.pop_front();
            memcpy(ep.send_buf(sb), payload.data(), payload.size());
            ep.post_send_msg(sb, payload.size());
        } else if (msg.op == OP_RDMA_WRITE_COMMIT) {
            srv->commit_addrs(conn, msg.remote_addrs);
        } else if (msg.op == OP_RDMA_READ) {
            if (msg.keys.size() != msg.remote_addrs.size() || msg.block_size <= 0) return;
            // The client pre-posted a bare recv for our WRITE_WITH_IMM; on
            // any failure answer with a zero-WR IMM carrying the status so
            // it fails fast instead of waiting out its 10 s timeout (the
            // reference just drops the request, infinistore.cpp:443-446).
            auto fail = [&](int status) {
                flow.submit({}, /*with_imm=*/true, static_cast<uint32_t>(status),
                            /*cookie=*/0);
            };
            auto held = std::make_shared<std::vector<Ref<BlockEntry>>>();
            if (!srv->collect_read_entries(msg.keys, held.get())) {
                WARN("verbs read: missing/uncommitted key");
                return fail(KEY_NOT_FOUND);
            }
            std::vector<WrDesc> wrs;
            wrs.reserve(held->size());
            for (size_t i = 0; i < held->size(); i++) {
                BlockEntry* e = (*held)[i].get();
                vf::MrInfo mr;
                if (!srv->vdrv_->lookup_region(e->ptr, &mr)) {
                    ERROR("verbs read: pool arena not registered");
                    return fail(SYSTEM_ERROR);
                }
                wrs.push_back({reinterpret_cast<uint64_t>(e->ptr), msg.remote_addrs[i],
                               static_cast<uint32_t>(msg.block_size), mr.lkey, msg.rkey});
            }
            uint32_t cookie = next_cookie++;
            if (!next_cookie) next_cookie = 1;
            reads.emplace(cookie, held);
            flow.submit(std::move(wrs), /*with_imm=*/true, /*imm=*/0, cookie);
        }
    }
};

bool Server::verbs_handshake(Conn* c, const std::vector<uint8_t>& body,
                             std::vector<uint8_t>* reply) {
    if (!vf::compiled_in() || c->verbs) return false;
    if (!vdrv_) {
        vf::Options o;
        o.dev_name = opt_.dev_name;
        o.ib_port = opt_.ib_port;
        o.roce = opt_.link_type != "IB";
        if (!vf::device_available(o)) return false;
        auto drv = std::make_unique<vf::Driver>();
        std::string err;
        if (!drv->init(o, &err)) {
            WARN("verbs driver init failed: %s", err.c_str());
            return false;
        }
        // Register every pool arena once (HBM arenas via dmabuf/peer-direct).
        // Held under vdrv_mu_ so an arena added by a concurrent pool
        // extension is registered by exactly one side (the extend thread
        // registers only when it observes vdrv_ set).
        std::lock_guard<std::mutex> lk(vdrv_mu_);
        bool mr_ok = true;
        for (auto& s : shards_) {
            s->for_each_arena([&](void* base, size_t sz, bool on_gpu) {
                mr_ok = mr_ok && drv->reg_region(base, sz, on_gpu, nullptr);
            });
        }
        if (!mr_ok) {
            WARN("pool MR registration failed; staying on the TCP fabric");
            return false;
        }
        vdrv_ = std::move(drv);
    }

    auto* peer = new VerbsPeer(this, c);
    vf::ConnInfo local{}, remote{};
    memcpy(&remote, body.data(), sizeof(remote));
    std::string err;
    if (!peer->ep.init(&local, &err) || !peer->ep.connect(remote, &err)) {
        WARN("verbs handshake failed: %s", err.c_str());
        delete peer;
        return false;
    }
    for (int i = 0; i < vf::kRecvBufs; i++) peer->ep.post_recv_buf(i);
    for (int i = 0; i < vf::kSendBufs; i++) peer->free_send.push_back(i);

    uv_poll_init(&c->owner->loop, &peer->poll, peer->ep.comp_fd());
    peer->poll.data = peer;
    uv_poll_start(&peer->poll, UV_READABLE, [](uv_poll_t* h, int status, int) {
        if (status < 0) return;
        auto* p = static_cast<Server::VerbsPeer*>(h->data);
        p->ep.drain([p](vf::Ev ev, uint64_t id, uint32_t imm, uint32_t len) {
            p->on_event(ev, id, imm, len);
        });
    });
    peer->polling = true;
    c->verbs = peer;

    reply->resize(4 + sizeof(vf::ConnInfo));
    memcpy(reply->data(), "VRBS", 4);
    memcpy(reply->data() + 4, &local, sizeof(local));
    INFO("verbs fabric established for client (qpn=%u)", local.qpn);
    return true;
}

void Server::verbs_teardown(Conn* c) {
    if (!c->verbs) return;
    VerbsPeer* peer = c->verbs;
    c->verbs = nullptr;
    if (peer->polling) {
        uv_poll_stop(&peer->poll);
        peer->poll.data = peer;
        uv_close(reinterpret_cast<uv_handle_t*>(&peer->poll), [](uv_handle_t* h) {
            delete static_cast<Server::VerbsPeer*>(h->data);
        });
    } else {
        delete peer;
    }
}

}  // namespace ifs
