// ibverbs RDMA transport for infinistore-amd.
//
// Same data-plane design as the reference: one RC QP per
// client connection, conn info exchanged over the TCP control plane,
// flatbuffers metadata in-band via IBV_WR_SEND into pre-posted 4 MB recv
// buffers, payloads as one-sided RDMA_WRITE chains (last WR signaled /
// WRITE_WITH_IMM at set boundaries, flow-controlled by WrFlow). GPU memory
// registers through the amdgpu dmabuf path (`ibv_reg_dmabuf_mr` on
// hipMalloc'd HBM, gpu::export_dmabuf) with plain `ibv_reg_mr` as the
// peer-direct fallback — the MI355X replacement for nv_peer_mem.
//
// Compiled fully only where rdma-core (<infiniband/verbs.h>) is present at
// build time; otherwise the module compiles to stubs and the TCP fabric
// serves the RDMA-semantics API. This build environment has no rdma-core
// and no RDMA NIC, so this module is syntax-checked against a mock verbs
// header (tests/mock_verbs) and NOT runtime-tested; treat as experimental.
#pragma once

#include <cstdint>
#include <functional>
#include <memory>
#include <string>
#include <vector>

#include "wr_flow.h"

#if defined(__has_include)
#if __has_include(<infiniband/verbs.h>)
#define IFS_HAVE_VERBS 1
#else
#define IFS_HAVE_VERBS 0
#endif
#else
#define IFS_HAVE_VERBS 0
#endif

namespace ifs {
namespace vf {

// Packed conn-exchange struct; field layout mirrors the reference's
// rdma_conn_info_t (protocol.h:85-91: qpn, psn, 16-byte gid, lid, mtu).
#pragma pack(push, 1)
struct ConnInfo {
    uint32_t qpn;
    uint32_t psn;
    uint8_t gid[16];
    uint16_t lid;
    uint32_t mtu;
};
#pragma pack(pop)
static_assert(sizeof(ConnInfo) == 30, "ConnInfo wire size");

constexpr size_t kMsgBufSize = kProtocolBufferSize;  // 4 MB protocol buffers
constexpr int kRecvBufs = 64;                        // MAX_RECV_WR role
constexpr int kSendBufs = 64;

struct Options {
    std::string dev_name;  // empty = first active device
    int ib_port = 1;
    bool roce = true;  // Ethernet/RoCEv2 (GID discovery) vs IB (lid)
};

// True when built against rdma-core.
bool compiled_in();
// True when an RDMA device with an active port exists at runtime.
bool device_available(const Options& opt);

struct MrInfo {
    void* addr = nullptr;
    size_t len = 0;
    uint32_t lkey = 0;
    uint32_t rkey = 0;
    void* handle = nullptr;  // ibv_mr*
};

// Process-wide verbs state: device context + PD + registered-region table
// (one per process, like the reference's init_rdma_context +
// pool-wide MRs, infinistore.cpp:806-871). Endpoints (QPs) share it, so
// pool arenas register once, not per client connection.
class Driver {
   public:
    Driver();
    ~Driver();
    bool init(const Options& opt, std::string* err);
    bool ready() const;

    // Register a region (pool arena / client tensor). Thread-safe; regions
    // are looked up by containment for lkey resolution.
    bool reg_region(void* addr, size_t len, bool device_mem, MrInfo* out);
    bool lookup_region(const void* ptr, MrInfo* out) const;

   private:
    friend class Endpoint;
    struct Impl;
    std::unique_ptr<Impl> impl_;
};

// Completion event kinds surfaced by Endpoint::poll.
enum class Ev { kSendDone, kRecvMsg, kRecvImm, kWriteDone, kError };
// cb(kind, wr_id, imm_or_len, byte_len)
using EventCb = std::function<void(Ev, uint64_t, uint32_t, uint32_t)>;

class Endpoint {
   public:
    explicit Endpoint(Driver& drv);
    ~Endpoint();
    Endpoint(const Endpoint&) = delete;
    Endpoint& operator=(const Endpoint&) = delete;

    // CQ/QP bring-up on the shared driver; fills the local ConnInfo.
    bool init(ConnInfo* local, std::string* err);
    // Transition the QP to RTS against the peer's info (MTU negotiated min).
    bool connect(const ConnInfo& remote, std::string* err);

    // Pre-registered message buffers (kMsgBufSize each).
    uint8_t* recv_buf(int i);
    uint8_t* send_buf(int i);
    bool post_recv_buf(int i);                    // wr_id = i (recv space)
    bool post_send_msg(int i, size_t len);        // wr_id = kSendWrBase + i
    bool post_recv_bare(uint64_t wr_id);          // zero-length (IMM wait)

    // One-sided write chain (WrFlow's PostFn); wr_id carries the cookie.
    bool post_write_chain(const WrChain& ch);

    // Completion plumbing: fd to poll, then drain() on readiness.
    int comp_fd() const;
    bool arm();                      // request next CQ event notification
    int drain(const EventCb& cb);    // ack + poll all; returns #CQEs or -1

    // wr_id encodings: sends carry the buffer index; write chains carry
    // (chain_len << 32 | cookie) so the CQ handler can meter WrFlow.
    static constexpr uint64_t kSendWrBase = 1ull << 61;
    static constexpr uint64_t kWriteWrBase = 1ull << 62;
    static uint64_t write_cookie(uint64_t payload) { return payload & 0xffffffffu; }
    static uint64_t write_chain_len(uint64_t payload) { return (payload >> 32) & 0xffffff; }

   private:
    struct Impl;
    std::unique_ptr<Impl> impl_;
};

}  // namespace vf
}  // namespace ifs
