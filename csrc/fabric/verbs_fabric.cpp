#include "verbs_fabric.h"

#include "../core/log.h"

#if IFS_HAVE_VERBS

#include <arpa/inet.h>
#include <fcntl.h>
#include <infiniband/verbs.h>
#include <unistd.h>

#include <algorithm>
#include <cstdio>
#include <cstring>
#include <mutex>
#include <random>

#include "../gpu/gpu.h"

namespace ifs {
namespace vf {

bool compiled_in() { return true; }

namespace {

ibv_device* find_device(const std::string& name) {
    int n = 0;
    ibv_device** list = ibv_get_device_list(&n);
    if (!list) return nullptr;
    ibv_device* found = nullptr;
    for (int i = 0; i < n; i++) {
        if (name.empty() || name == ibv_get_device_name(list[i])) {
            found = list[i];
            break;
        }
    }
    // The list must stay alive while the device is open; leaked on purpose
    // (one per process bring-up).
    return found;
}

// RoCE v2 GID discovery via sysfs (role of the reference's vendored
// ibv_helper.cpp:107-130): pick the GID index whose type is RoCE v2,
// preferring IPv4-mapped GIDs.
int find_roce_v2_gid(ibv_context* ctx, const std::string& dev, int port, ibv_gid* out) {
    int best = -1;
    bool best_v4 = false;
    for (int i = 0; i < 16; i++) {
        char path[256];
        snprintf(path, sizeof(path), "/sys/class/infiniband/%s/ports/%d/gid_attrs/types/%d",
                 dev.c_str(), port, i);
        FILE* f = fopen(path, "r");
        if (!f) continue;
        char buf[64] = {0};
        size_t r = fread(buf, 1, sizeof(buf) - 1, f);
        fclose(f);
        (void)r;
        if (!strstr(buf, "RoCE v2")) continue;
        ibv_gid gid;
        if (ibv_query_gid(ctx, static_cast<uint8_t>(port), i, &gid) != 0) continue;
        bool zero = true;
        for (int b = 0; b < 16; b++) zero &= gid.raw[b] == 0;
        if (zero) continue;
        bool v4 = gid.raw[10] == 0xff && gid.raw[11] == 0xff;  // ::ffff:a.b.c.d
        if (best < 0 || (v4 && !best_v4)) {
            best = i;
            best_v4 = v4;
            *out = gid;
        }
    }
    return best;
}

}  // namespace

bool device_available(const Options& opt) {
    ibv_device* dev = find_device(opt.dev_name);
    if (!dev) return false;
    ibv_context* ctx = ibv_open_device(dev);
    if (!ctx) return false;
    ibv_port_attr pa;
    bool ok = ibv_query_port(ctx, static_cast<uint8_t>(opt.ib_port), &pa) == 0 &&
              pa.state == IBV_PORT_ACTIVE;
    ibv_close_device(ctx);
    return ok;
}

// ---------------------------------------------------------------------------
// Driver: shared device context + PD + region table
// ---------------------------------------------------------------------------
struct Driver::Impl {
    Options opt;
    ibv_context* ctx = nullptr;
    ibv_pd* pd = nullptr;
    ibv_port_attr port_attr{};
    ibv_gid gid{};
    int gid_idx = -1;
    std::string dev_name;
    bool ready = false;

    mutable std::mutex mr_mu;
    std::vector<MrInfo> regions;

    ~Impl() {
        for (auto& mr : regions)
            if (mr.handle) ibv_dereg_mr(static_cast<ibv_mr*>(mr.handle));
        if (pd) ibv_dealloc_pd(pd);
        if (ctx) ibv_close_device(ctx);
    }
};

Driver::Driver() : impl_(new Impl()) {}
Driver::~Driver() = default;
bool Driver::ready() const { return impl_->ready; }

bool Driver::init(const Options& opt, std::string* err) {
    auto& im = *impl_;
    im.opt = opt;
    ibv_device* dev = find_device(opt.dev_name);
    if (!dev) {
        *err = "no RDMA device '" + opt.dev_name + "'";
        return false;
    }
    im.dev_name = ibv_get_device_name(dev);
    im.ctx = ibv_open_device(dev);
    if (!im.ctx) {
        *err = "ibv_open_device failed";
        return false;
    }
    if (ibv_query_port(im.ctx, static_cast<uint8_t>(opt.ib_port), &im.port_attr) != 0 ||
        im.port_attr.state != IBV_PORT_ACTIVE) {
        *err = "port not active";
        return false;
    }
    if (im.port_attr.link_layer == IBV_LINK_LAYER_ETHERNET) {
        im.gid_idx = find_roce_v2_gid(im.ctx, im.dev_name, opt.ib_port, &im.gid);
        if (im.gid_idx < 0) {
            *err = "no RoCE v2 GID";
            return false;
        }
    }
    im.pd = ibv_alloc_pd(im.ctx);
    if (!im.pd) {
        *err = "ibv_alloc_pd failed";
        return false;
    }
    im.ready = true;
    INFO("verbs driver up: dev=%s port=%d link=%s", im.dev_name.c_str(), opt.ib_port,
         im.port_attr.link_layer == IBV_LINK_LAYER_ETHERNET ? "RoCEv2" : "IB");
    return true;
}

bool Driver::reg_region(void* addr, size_t len, bool device_mem, MrInfo* out) {
    auto& im = *impl_;
    if (!im.ready) return false;
    int access = IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_WRITE | IBV_ACCESS_REMOTE_READ;
    ibv_mr* mr = nullptr;
    if (device_mem) {
        // amdgpu dmabuf path — the MI355X replacement for nv_peer_mem.
        uint64_t off = 0;
        int fd = gpu::export_dmabuf(addr, len, &off);
        if (fd >= 0) {
            mr = ibv_reg_dmabuf_mr(im.pd, off, len, reinterpret_cast<uint64_t>(addr), fd,
                                   access);
            close(fd);
        }
        if (!mr) mr = ibv_reg_mr(im.pd, addr, len, access);  // peer-direct fallback
    } else {
        mr = ibv_reg_mr(im.pd, addr, len, access);
    }
    if (!mr) {
        ERROR("ibv MR registration failed for %p (%zu bytes, device=%d)", addr, len,
              device_mem);
        return false;
    }
    MrInfo info{addr, len, mr->lkey, mr->rkey, mr};
    {
        std::lock_guard<std::mutex> lk(im.mr_mu);
        im.regions.push_back(info);
    }
    if (out) *out = info;
    return true;
}

bool Driver::lookup_region(const void* ptr, MrInfo* out) const {
    auto& im = *impl_;
    auto p = reinterpret_cast<uintptr_t>(ptr);
    std::lock_guard<std::mutex> lk(im.mr_mu);
    for (auto& mr : im.regions) {
        auto b = reinterpret_cast<uintptr_t>(mr.addr);
        if (p >= b && p < b + mr.len) {
            *out = mr;
            return true;
        }
    }
    return false;
}

// ---------------------------------------------------------------------------
// Endpoint: one RC QP on the shared driver
// ---------------------------------------------------------------------------
struct Endpoint::Impl {
    Driver* drv = nullptr;
    ibv_comp_channel* channel = nullptr;
    ibv_cq* cq = nullptr;
    ibv_qp* qp = nullptr;
    uint32_t psn = 0;
    ibv_mtu active_mtu = IBV_MTU_1024;

    uint8_t* msg_region = nullptr;  // recv bufs then send bufs, one MR
    ibv_mr* msg_mr = nullptr;

    ~Impl() {
        if (qp) ibv_destroy_qp(qp);
        if (cq) ibv_destroy_cq(cq);
        if (channel) ibv_destroy_comp_channel(channel);
        if (msg_mr) ibv_dereg_mr(msg_mr);
        free(msg_region);
    }
};

Endpoint::Endpoint(Driver& drv) : impl_(new Impl()) { impl_->drv = &drv; }
Endpoint::~Endpoint() = default;

bool Endpoint::init(ConnInfo* local, std::string* err) {
    auto& im = *impl_;
    auto& d = *im.drv->impl_;
    if (!d.ready) {
        *err = "driver not initialized";
        return false;
    }
    im.channel = ibv_create_comp_channel(d.ctx);
    if (!im.channel) {
        *err = "comp channel failed";
        return false;
    }
    im.cq = ibv_create_cq(d.ctx, 8192 + kRecvBufs, nullptr, im.channel, 0);
    if (!im.cq) {
        *err = "cq failed";
        return false;
    }
    ibv_qp_init_attr qia{};
    qia.send_cq = im.cq;
    qia.recv_cq = im.cq;
    qia.qp_type = IBV_QPT_RC;
    qia.cap.max_send_wr = 8192;  // MAX_SEND_WR role
    qia.cap.max_recv_wr = kRecvBufs + 16;
    qia.cap.max_send_sge = 1;
    qia.cap.max_recv_sge = 1;
    im.qp = ibv_create_qp(d.pd, &qia);
    if (!im.qp) {
        *err = "qp create failed";
        return false;
    }
    ibv_qp_attr a{};
    a.qp_state = IBV_QPS_INIT;
    a.pkey_index = 0;
    a.port_num = static_cast<uint8_t>(d.opt.ib_port);
    a.qp_access_flags =
        IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_WRITE | IBV_ACCESS_REMOTE_READ;
    if (ibv_modify_qp(im.qp, &a,
                      IBV_QP_STATE | IBV_QP_PKEY_INDEX | IBV_QP_PORT | IBV_QP_ACCESS_FLAGS) !=
        0) {
        *err = "modify INIT failed";
        return false;
    }
    size_t region = kMsgBufSize * (kRecvBufs + kSendBufs);
    if (posix_memalign(reinterpret_cast<void**>(&im.msg_region), 4096, region) != 0) {
        *err = "msg region alloc failed";
        return false;
    }
    im.msg_mr = ibv_reg_mr(d.pd, im.msg_region, region, IBV_ACCESS_LOCAL_WRITE);
    if (!im.msg_mr) {
        *err = "msg MR reg failed";
        return false;
    }
    std::random_device rd;
    im.psn = rd() & 0xffffff;
    im.active_mtu = d.port_attr.active_mtu;

    memset(local, 0, sizeof(*local));
    local->qpn = im.qp->qp_num;
    local->psn = im.psn;
    memcpy(local->gid, d.gid.raw, 16);
    local->lid = d.port_attr.lid;
    local->mtu = static_cast<uint32_t>(im.active_mtu);
    return true;
}

bool Endpoint::connect(const ConnInfo& remote, std::string* err) {
    auto& im = *impl_;
    auto& d = *im.drv->impl_;
    ibv_qp_attr a{};
    a.qp_state = IBV_QPS_RTR;
    // MTU negotiation: min of both ends (reference infinistore.cpp:946-954).
    a.path_mtu = static_cast<ibv_mtu>(
        std::min<uint32_t>(static_cast<uint32_t>(im.active_mtu), remote.mtu));
    a.dest_qp_num = remote.qpn;
    a.rq_psn = remote.psn;
    a.max_dest_rd_atomic = 1;
    a.min_rnr_timer = 12;
    a.ah_attr.port_num = static_cast<uint8_t>(d.opt.ib_port);
    if (d.port_attr.link_layer == IBV_LINK_LAYER_ETHERNET) {
        a.ah_attr.is_global = 1;
        memcpy(a.ah_attr.grh.dgid.raw, remote.gid, 16);
        a.ah_attr.grh.sgid_index = static_cast<uint8_t>(d.gid_idx);
        a.ah_attr.grh.hop_limit = 64;
    } else {
        a.ah_attr.is_global = 0;
        a.ah_attr.dlid = remote.lid;
    }
    if (ibv_modify_qp(im.qp, &a,
                      IBV_QP_STATE | IBV_QP_AV | IBV_QP_PATH_MTU | IBV_QP_DEST_QPN |
                          IBV_QP_RQ_PSN | IBV_QP_MAX_DEST_RD_ATOMIC | IBV_QP_MIN_RNR_TIMER) !=
        0) {
        *err = "modify RTR failed";
        return false;
    }
    ibv_qp_attr b{};
    b.qp_state = IBV_QPS_RTS;
    b.timeout = 14;
    b.retry_cnt = 7;
    b.rnr_retry = 7;
    b.sq_psn = im.psn;
    b.max_rd_atomic = 1;
    if (ibv_modify_qp(im.qp, &b,
                      IBV_QP_STATE | IBV_QP_TIMEOUT | IBV_QP_RETRY_CNT | IBV_QP_RNR_RETRY |
                          IBV_QP_SQ_PSN | IBV_QP_MAX_QP_RD_ATOMIC) != 0) {
        *err = "modify RTS failed";
        return false;
    }
    return arm();
}

uint8_t* Endpoint::recv_buf(int i) { return impl_->msg_region + kMsgBufSize * i; }
uint8_t* Endpoint::send_buf(int i) {
    return impl_->msg_region + kMsgBufSize * (kRecvBufs + i);
}

bool Endpoint::post_recv_buf(int i) {
    auto& im = *impl_;
    ibv_sge sge{reinterpret_cast<uint64_t>(recv_buf(i)), static_cast<uint32_t>(kMsgBufSize),
                im.msg_mr->lkey};
    ibv_recv_wr wr{};
    wr.wr_id = static_cast<uint64_t>(i);
    wr.sg_list = &sge;
    wr.num_sge = 1;
    ibv_recv_wr* bad = nullptr;
    return ibv_post_recv(im.qp, &wr, &bad) == 0;
}

bool Endpoint::post_recv_bare(uint64_t wr_id) {
    auto& im = *impl_;
    ibv_recv_wr wr{};
    wr.wr_id = wr_id;
    wr.sg_list = nullptr;
    wr.num_sge = 0;
    ibv_recv_wr* bad = nullptr;
    return ibv_post_recv(im.qp, &wr, &bad) == 0;
}

bool Endpoint::post_send_msg(int i, size_t len) {
    auto& im = *impl_;
    ibv_sge sge{reinterpret_cast<uint64_t>(send_buf(i)), static_cast<uint32_t>(len),
                im.msg_mr->lkey};
    ibv_send_wr wr{};
    wr.wr_id = kSendWrBase + static_cast<uint64_t>(i);
    wr.sg_list = &sge;
    wr.num_sge = 1;
    wr.opcode = IBV_WR_SEND;
    wr.send_flags = IBV_SEND_SIGNALED;
    ibv_send_wr* bad = nullptr;
    return ibv_post_send(im.qp, &wr, &bad) == 0;
}

bool Endpoint::post_write_chain(const WrChain& ch) {
    auto& im = *impl_;
    size_t n = ch.wrs.size();
    std::vector<ibv_send_wr> wrs(std::max<size_t>(n, 1));
    std::vector<ibv_sge> sges(std::max<size_t>(n, 1));
    if (n == 0) {
        // bare IMM (zero-length write) marking a set boundary
        ibv_send_wr& wr = wrs[0];
        memset(&wr, 0, sizeof(wr));
        wr.wr_id = kWriteWrBase | (ch.signal_cookie & 0xffffffffu);
        wr.opcode = IBV_WR_RDMA_WRITE_WITH_IMM;
        wr.imm_data = htonl(ch.imm_data);
        wr.send_flags = IBV_SEND_SIGNALED;
        ibv_send_wr* bad = nullptr;
        return ibv_post_send(im.qp, &wr, &bad) == 0;
    }
    for (size_t i = 0; i < n; i++) {
        const WrDesc& d = ch.wrs[i];
        sges[i] = ibv_sge{d.local_addr, d.len, d.lkey};
        ibv_send_wr& wr = wrs[i];
        memset(&wr, 0, sizeof(wr));
        wr.sg_list = &sges[i];
        wr.num_sge = 1;
        wr.opcode = IBV_WR_RDMA_WRITE;
        wr.wr.rdma.remote_addr = d.remote_addr;
        wr.wr.rdma.rkey = d.rkey;
        wr.next = (i + 1 < n) ? &wrs[i + 1] : nullptr;
    }
    ibv_send_wr& last = wrs[n - 1];
    last.send_flags = IBV_SEND_SIGNALED;
    last.wr_id = kWriteWrBase | (static_cast<uint64_t>(n) << 32) |
                 (ch.signal_cookie & 0xffffffffu);
    if (ch.with_imm) {
        last.opcode = IBV_WR_RDMA_WRITE_WITH_IMM;
        last.imm_data = htonl(ch.imm_data);
    }
    ibv_send_wr* bad = nullptr;
    return ibv_post_send(im.qp, &wrs[0], &bad) == 0;
}

int Endpoint::comp_fd() const { return impl_->channel ? impl_->channel->fd : -1; }

bool Endpoint::arm() { return ibv_req_notify_cq(impl_->cq, 0) == 0; }

int Endpoint::drain(const EventCb& cb) {
    auto& im = *impl_;
    ibv_cq* ev_cq = nullptr;
    void* ev_ctx = nullptr;
    if (ibv_get_cq_event(im.channel, &ev_cq, &ev_ctx) == 0) ibv_ack_cq_events(ev_cq, 1);
    if (!arm()) return -1;
    int total = 0;
    ibv_wc wc[16];
    for (;;) {
        int n = ibv_poll_cq(im.cq, 16, wc);
        if (n < 0) return -1;
        if (n == 0) break;
        for (int i = 0; i < n; i++) {
            if (wc[i].status != IBV_WC_SUCCESS) {
                ERROR("CQE error: %s (wr_id=%llx)", ibv_wc_status_str(wc[i].status),
                      (unsigned long long)wc[i].wr_id);
                cb(Ev::kError, wc[i].wr_id, 0, 0);
                continue;
            }
            switch (wc[i].opcode) {
                case IBV_WC_SEND:
                    cb(Ev::kSendDone, wc[i].wr_id - kSendWrBase, 0, 0);
                    break;
                case IBV_WC_RECV:
                    cb(Ev::kRecvMsg, wc[i].wr_id, 0, wc[i].byte_len);
                    break;
                case IBV_WC_RECV_RDMA_WITH_IMM:
                    cb(Ev::kRecvImm, wc[i].wr_id, ntohl(wc[i].imm_data), wc[i].byte_len);
                    break;
                case IBV_WC_RDMA_WRITE:
                    cb(Ev::kWriteDone, wc[i].wr_id & ~kWriteWrBase, 0, 0);
                    break;
                default:
                    break;
            }
        }
        total += n;
    }
    return total;
}

}  // namespace vf
}  // namespace ifs

#else  // !IFS_HAVE_VERBS — stubs (the TCP fabric serves the API)

namespace ifs {
namespace vf {

bool compiled_in() { return false; }
bool device_available(const Options&) { return false; }

struct Driver::Impl {};
Driver::Driver() = default;
Driver::~Driver() = default;
bool Driver::ready() const { return false; }
bool Driver::init(const Options&, std::string* err) {
    *err = "built without rdma-core";
    return false;
}
bool Driver::reg_region(void*, size_t, bool, MrInfo*) { return false; }
bool Driver::lookup_region(const void*, MrInfo*) const { return false; }

struct Endpoint::Impl {};
Endpoint::Endpoint(Driver&) : impl_(new Impl()) {}
Endpoint::~Endpoint() = default;
bool Endpoint::init(ConnInfo*, std::string* err) {
    *err = "built without rdma-core";
    return false;
}
bool Endpoint::connect(const ConnInfo&, std::string* err) {
    *err = "built without rdma-core";
    return false;
}
uint8_t* Endpoint::recv_buf(int) { return nullptr; }
uint8_t* Endpoint::send_buf(int) { return nullptr; }
bool Endpoint::post_recv_buf(int) { return false; }
bool Endpoint::post_recv_bare(uint64_t) { return false; }
bool Endpoint::post_send_msg(int, size_t) { return false; }
bool Endpoint::post_write_chain(const WrChain&) { return false; }
int Endpoint::comp_fd() const { return -1; }
bool Endpoint::arm() { return false; }
int Endpoint::drain(const EventCb&) { return -1; }

}  // namespace vf
}  // namespace ifs

#endif  // IFS_HAVE_VERBS
