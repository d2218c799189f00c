// In-process loopback implementation of the mock <infiniband/verbs.h> —
// turns the compile-only harness into a RUNTIME harness: both fabric ends
// live in one test process, so "remote" addresses are real pointers and
// RDMA_WRITE is a memcpy into the target MR. QPs pair through a global
// qp_num registry (ibv_modify_qp(..., IBV_QP_DEST_QPN) wires the peer);
// SEND consumes the peer's posted recv; WRITE_WITH_IMM additionally
// delivers an IBV_WC_RECV_RDMA_WITH_IMM completion. Completion channels
// are pipes so the server's uv_poll and the client's blocking
// ibv_get_cq_event work unchanged.
//
// Test infrastructure only (linked by scripts/san_build.py mockverbs mode);
// never shipped. The port reports LINK_LAYER_INFINIBAND so the production
// code takes the LID addressing path and never touches sysfs.

#include <infiniband/verbs.h>
#include <pthread.h>
#include <string.h>
#include <unistd.h>

#include <atomic>
#include <deque>
#include <map>
#include <mutex>
#include <vector>

// Completes the header's forward declaration — must be at global scope.
struct ibv_device {
    int dummy;
};

namespace {

struct CqImpl;

struct ChannelImpl {
    ibv_comp_channel pub;  // .fd = read end of the pipe
    int wfd = -1;
};

struct CqImpl {
    ibv_cq pub;
    std::mutex mu;
    std::deque<ibv_wc> q;
    ChannelImpl* ch = nullptr;
    void* cq_context = nullptr;
    bool armed = false;

    void push(const ibv_wc& wc) {
        bool fire = false;
        {
            std::lock_guard<std::mutex> lk(mu);
            q.push_back(wc);
            if (armed) {
                armed = false;
                fire = true;
            }
        }
        if (fire && ch) {
            char b = 1;
            ssize_t r = write(ch->wfd, &b, 1);
            (void)r;
        }
    }
};

struct RecvSlot {
    uint64_t wr_id;
    uint64_t addr;
    uint32_t length;
};

struct QpImpl {
    ibv_qp pub;
    CqImpl* send_cq = nullptr;
    CqImpl* recv_cq = nullptr;
    int sq_sig_all = 0;
    uint32_t dest_qpn = 0;
    std::mutex mu;
    std::deque<RecvSlot> recvq;
};

std::mutex g_mu;
std::map<uint32_t, QpImpl*> g_qps;
std::atomic<uint32_t> g_next_qpn{100};
std::atomic<uint32_t> g_next_key{1000};

// Registered-MR table: like a real NIC, every RDMA_WRITE's rkey must name an
// MR that covers [remote_addr, remote_addr+len) with REMOTE_WRITE access, and
// every SGE's lkey an MR covering the local range. The round-1 mock skipped
// this, which let a placeholder-rkey bug (allocate responses carrying
// device+1 instead of the pool MR's rkey) pass the whole loopback suite while
// guaranteeing remote-access errors on real hardware.
struct MockMr {
    uint64_t addr;
    size_t len;
    uint32_t lkey;
    uint32_t rkey;
    int access;
};
std::mutex g_mr_mu;
std::map<uint32_t, MockMr> g_mrs_by_lkey;  // rkey = lkey + 1 (one entry per MR)

bool check_rkey(uint32_t rkey, uint64_t addr, size_t len) {
    if (len == 0) return true;  // zero-length writes skip rkey checks (mlx behavior)
    std::lock_guard<std::mutex> lk(g_mr_mu);
    for (auto& [k, mr] : g_mrs_by_lkey) {
        if (mr.rkey == rkey && addr >= mr.addr && addr + len <= mr.addr + mr.len)
            return (mr.access & IBV_ACCESS_REMOTE_WRITE) != 0;
    }
    return false;
}

bool check_lkeys(const ibv_send_wr* wr) {
    std::lock_guard<std::mutex> lk(g_mr_mu);
    for (int i = 0; i < wr->num_sge; i++) {
        const ibv_sge& s = wr->sg_list[i];
        if (s.length == 0) continue;
        auto it = g_mrs_by_lkey.find(s.lkey);
        if (it == g_mrs_by_lkey.end()) return false;
        const MockMr& mr = it->second;
        if (s.addr < mr.addr || s.addr + s.length > mr.addr + mr.len) return false;
    }
    return true;
}

// One fake device.
ibv_device g_device;
ibv_device* g_list[2] = {&g_device, nullptr};
ibv_context g_ctx;
ibv_pd g_pd;

QpImpl* lookup_qp(uint32_t qpn) {
    std::lock_guard<std::mutex> lk(g_mu);
    auto it = g_qps.find(qpn);
    return it == g_qps.end() ? nullptr : it->second;
}

size_t copy_sges(const ibv_send_wr* wr, uint8_t* dst, size_t cap) {
    size_t off = 0;
    for (int i = 0; i < wr->num_sge; i++) {
        const ibv_sge& s = wr->sg_list[i];
        size_t n = s.length;
        if (off + n > cap) n = cap > off ? cap - off : 0;
        memcpy(dst + off, reinterpret_cast<const void*>(s.addr), n);
        off += n;
    }
    return off;
}

size_t sge_len(const ibv_send_wr* wr) {
    size_t n = 0;
    for (int i = 0; i < wr->num_sge; i++) n += wr->sg_list[i].length;
    return n;
}

}  // namespace

extern "C" {

ibv_device** ibv_get_device_list(int* num) {
    if (num) *num = 1;
    return g_list;
}
void ibv_free_device_list(ibv_device**) {}
const char* ibv_get_device_name(ibv_device*) { return "mock0"; }
ibv_context* ibv_open_device(ibv_device*) { return &g_ctx; }
int ibv_close_device(ibv_context*) { return 0; }

int ibv_query_port(ibv_context*, uint8_t, ibv_port_attr* attr) {
    memset(attr, 0, sizeof(*attr));
    attr->state = IBV_PORT_ACTIVE;
    attr->max_mtu = IBV_MTU_4096;
    attr->active_mtu = IBV_MTU_4096;
    attr->lid = 1;
    attr->link_layer = IBV_LINK_LAYER_INFINIBAND;  // LID path: no sysfs GIDs
    return 0;
}

int ibv_query_gid(ibv_context*, uint8_t, int, union ibv_gid* gid) {
    memset(gid->raw, 0, 16);
    gid->raw[15] = 1;
    return 0;
}

ibv_pd* ibv_alloc_pd(ibv_context*) { return &g_pd; }
int ibv_dealloc_pd(ibv_pd*) { return 0; }

ibv_comp_channel* ibv_create_comp_channel(ibv_context*) {
    int fds[2];
    if (pipe(fds) != 0) return nullptr;
    auto* c = new ChannelImpl();
    c->pub.fd = fds[0];
    c->wfd = fds[1];
    return &c->pub;
}
int ibv_destroy_comp_channel(ibv_comp_channel* ch) {
    auto* c = reinterpret_cast<ChannelImpl*>(ch);
    close(c->pub.fd);
    close(c->wfd);
    delete c;
    return 0;
}

ibv_cq* ibv_create_cq(ibv_context*, int, void* cq_context, ibv_comp_channel* ch, int) {
    auto* cq = new CqImpl();
    cq->ch = reinterpret_cast<ChannelImpl*>(ch);
    cq->cq_context = cq_context;
    return &cq->pub;
}
int ibv_destroy_cq(ibv_cq* cq) {
    delete reinterpret_cast<CqImpl*>(cq);
    return 0;
}

ibv_qp* ibv_create_qp(ibv_pd*, ibv_qp_init_attr* attr) {
    auto* qp = new QpImpl();
    qp->pub.qp_num = g_next_qpn.fetch_add(1);
    qp->send_cq = reinterpret_cast<CqImpl*>(attr->send_cq);
    qp->recv_cq = reinterpret_cast<CqImpl*>(attr->recv_cq);
    qp->sq_sig_all = attr->sq_sig_all;
    std::lock_guard<std::mutex> lk(g_mu);
    g_qps[qp->pub.qp_num] = qp;
    return &qp->pub;
}
int ibv_destroy_qp(ibv_qp* qp) {
    auto* q = reinterpret_cast<QpImpl*>(qp);
    {
        std::lock_guard<std::mutex> lk(g_mu);
        g_qps.erase(q->pub.qp_num);
    }
    delete q;
    return 0;
}

int ibv_modify_qp(ibv_qp* qp, ibv_qp_attr* attr, int mask) {
    auto* q = reinterpret_cast<QpImpl*>(qp);
    if (mask & IBV_QP_DEST_QPN) q->dest_qpn = attr->dest_qp_num;
    return 0;
}

ibv_mr* ibv_reg_mr(ibv_pd*, void* addr, size_t len, int access) {
    auto* mr = new ibv_mr();
    mr->lkey = g_next_key.fetch_add(2);
    mr->rkey = mr->lkey + 1;
    {
        std::lock_guard<std::mutex> lk(g_mr_mu);
        g_mrs_by_lkey[mr->lkey] =
            MockMr{reinterpret_cast<uint64_t>(addr), len, mr->lkey, mr->rkey, access};
    }
    return mr;
}
ibv_mr* ibv_reg_dmabuf_mr(ibv_pd*, uint64_t, size_t, uint64_t, int, int) {
    return nullptr;  // force the peer-direct/plain fallback in tests
}
int ibv_dereg_mr(ibv_mr* mr) {
    {
        std::lock_guard<std::mutex> lk(g_mr_mu);
        g_mrs_by_lkey.erase(mr->lkey);
    }
    delete mr;
    return 0;
}

int ibv_post_recv(ibv_qp* qp, ibv_recv_wr* wr, ibv_recv_wr** bad) {
    auto* q = reinterpret_cast<QpImpl*>(qp);
    std::lock_guard<std::mutex> lk(q->mu);
    for (; wr; wr = wr->next) {
        RecvSlot s{wr->wr_id, 0, 0};
        if (wr->num_sge > 0) {
            s.addr = wr->sg_list[0].addr;
            s.length = wr->sg_list[0].length;
        }
        q->recvq.push_back(s);
    }
    if (bad) *bad = nullptr;
    return 0;
}

int ibv_post_send(ibv_qp* qp, ibv_send_wr* wr, ibv_send_wr** bad) {
    auto* q = reinterpret_cast<QpImpl*>(qp);
    for (; wr; wr = wr->next) {
        QpImpl* peer = lookup_qp(q->dest_qpn);
        if (!peer) {
            if (bad) *bad = wr;
            return 1;
        }
        ibv_wc lwc{};
        lwc.wr_id = wr->wr_id;
        lwc.status = IBV_WC_SUCCESS;
        lwc.qp_num = q->pub.qp_num;
        // lkey validation applies to every opcode's local SGEs.
        if (!check_lkeys(wr)) {
            lwc.status = IBV_WC_LOC_PROT_ERR;
            lwc.opcode = wr->opcode == IBV_WR_SEND ? IBV_WC_SEND : IBV_WC_RDMA_WRITE;
            q->send_cq->push(lwc);  // errored WRs always complete
            continue;
        }
        switch (wr->opcode) {
            case IBV_WR_SEND: {
                RecvSlot slot{};
                {
                    std::lock_guard<std::mutex> lk(peer->mu);
                    if (peer->recvq.empty()) {
                        if (bad) *bad = wr;
                        return 1;  // RNR in real life
                    }
                    slot = peer->recvq.front();
                    peer->recvq.pop_front();
                }
                if (sge_len(wr) > slot.length) {
                    // message longer than the posted recv buffer: receiver
                    // completes with a local-length error, nothing is copied.
                    ibv_wc rwc{};
                    rwc.wr_id = slot.wr_id;
                    rwc.status = IBV_WC_LOC_LEN_ERR;
                    rwc.opcode = IBV_WC_RECV;
                    rwc.qp_num = peer->pub.qp_num;
                    peer->recv_cq->push(rwc);
                    lwc.opcode = IBV_WC_SEND;
                    lwc.status = IBV_WC_REM_OP_ERR;
                    q->send_cq->push(lwc);
                    continue;
                }
                size_t n =
                    copy_sges(wr, reinterpret_cast<uint8_t*>(slot.addr), slot.length);
                ibv_wc rwc{};
                rwc.wr_id = slot.wr_id;
                rwc.status = IBV_WC_SUCCESS;
                rwc.opcode = IBV_WC_RECV;
                rwc.byte_len = static_cast<uint32_t>(n);
                rwc.qp_num = peer->pub.qp_num;
                peer->recv_cq->push(rwc);
                lwc.opcode = IBV_WC_SEND;
                break;
            }
            case IBV_WR_RDMA_WRITE:
            case IBV_WR_RDMA_WRITE_WITH_IMM: {
                if (!check_rkey(wr->wr.rdma.rkey, wr->wr.rdma.remote_addr, sge_len(wr))) {
                    lwc.status = IBV_WC_REM_ACCESS_ERR;
                    lwc.opcode = IBV_WC_RDMA_WRITE;
                    q->send_cq->push(lwc);
                    continue;
                }
                size_t n = copy_sges(wr, reinterpret_cast<uint8_t*>(wr->wr.rdma.remote_addr),
                                     sge_len(wr));
                if (wr->opcode == IBV_WR_RDMA_WRITE_WITH_IMM) {
                    RecvSlot slot{};
                    {
                        std::lock_guard<std::mutex> lk(peer->mu);
                        if (!peer->recvq.empty()) {
                            slot = peer->recvq.front();
                            peer->recvq.pop_front();
                        }
                    }
                    ibv_wc rwc{};
                    rwc.wr_id = slot.wr_id;
                    rwc.status = IBV_WC_SUCCESS;
                    rwc.opcode = IBV_WC_RECV_RDMA_WITH_IMM;
                    rwc.imm_data = wr->imm_data;
                    rwc.byte_len = static_cast<uint32_t>(n);
                    rwc.qp_num = peer->pub.qp_num;
                    peer->recv_cq->push(rwc);
                }
                lwc.opcode = IBV_WC_RDMA_WRITE;
                break;
            }
            default:
                if (bad) *bad = wr;
                return 1;
        }
        if (q->sq_sig_all || (wr->send_flags & IBV_SEND_SIGNALED)) q->send_cq->push(lwc);
    }
    if (bad) *bad = nullptr;
    return 0;
}

int ibv_req_notify_cq(ibv_cq* cq, int) {
    auto* c = reinterpret_cast<CqImpl*>(cq);
    bool fire = false;
    {
        std::lock_guard<std::mutex> lk(c->mu);
        if (!c->q.empty()) {
            fire = true;  // completions already queued: fire immediately
        } else {
            c->armed = true;
        }
    }
    if (fire && c->ch) {
        char b = 1;
        ssize_t r = write(c->ch->wfd, &b, 1);
        (void)r;
    }
    return 0;
}

int ibv_get_cq_event(ibv_comp_channel* ch, ibv_cq** cq, void** cq_context) {
    auto* c = reinterpret_cast<ChannelImpl*>(ch);
    char b;
    if (read(c->pub.fd, &b, 1) != 1) return 1;
    // One CQ per channel in this codebase; the caller gets it back through
    // the Endpoint, so a null is acceptable only if never dereferenced —
    // return the CQ the channel saw last via a registry-free trick: the
    // caller of this codebase always knows its CQ; we stash nothing.
    if (cq) *cq = nullptr;
    if (cq_context) *cq_context = nullptr;
    return 0;
}
void ibv_ack_cq_events(ibv_cq*, unsigned int) {}

int ibv_poll_cq(ibv_cq* cq, int num, ibv_wc* wc) {
    auto* c = reinterpret_cast<CqImpl*>(cq);
    std::lock_guard<std::mutex> lk(c->mu);
    int n = 0;
    while (n < num && !c->q.empty()) {
        wc[n++] = c->q.front();
        c->q.pop_front();
    }
    return n;
}

const char* ibv_wc_status_str(ibv_wc_status s) {
    return s == IBV_WC_SUCCESS ? "success" : "error";
}

}  // extern "C"
