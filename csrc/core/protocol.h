// Wire protocol for infinistore-amd.
//
// Wire-compatible with the reference store's protocol
// (cf. /root/reference/src/protocol.h:39-71 for the 9-byte header, op chars
// and status codes, and /root/reference/src/*.fbs for the four flatbuffers
// tables) — reimplemented from scratch on top of the clean-room flatbuffers
// core in wire.h.
//
// Extensions over the reference (all additive):
//  * LocalMetaRequest gains field id 4 `base_offset:ulong` — the byte offset
//    of the tensor pointer inside its IPC-exported allocation, which lets
//    clients use tensors that are not at the base of a hipMalloc allocation
//    (the reference requires PYTORCH_NO_CUDA_MEMORY_CACHING instead).
//  * New ops for the TCP data fabric (RDMA-semantics emulation when no
//    RDMA NIC is present): OP_TCP_PUT 'P', OP_TCP_GET 'G'.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

#include "wire.h"

namespace ifs {

// ---- framing --------------------------------------------------------------
constexpr uint32_t kMagic = 0xdeadbeef;

#pragma pack(push, 1)
struct Header {
    uint32_t magic;
    char op;
    uint32_t body_size;
};
#pragma pack(pop)
static_assert(sizeof(Header) == 9, "header must be 9 bytes");

// ---- ops ------------------------------------------------------------------
constexpr char OP_R = 'R';                  // local-GPU read
constexpr char OP_W = 'W';                  // local-GPU write
constexpr char OP_SYNC = 'S';               // local sync (inflight count)
constexpr char OP_RDMA_EXCHANGE = 'E';      // fabric negotiation
constexpr char OP_RDMA_ALLOCATE = 'D';      // allocate blocks for remote write
constexpr char OP_RDMA_READ = 'A';          // remote read request
constexpr char OP_RDMA_WRITE_COMMIT = 'T';  // commit written blocks
constexpr char OP_CHECK_EXIST = 'C';
constexpr char OP_GET_MATCH_LAST_IDX = 'M';
// Extensions (TCP data fabric):
constexpr char OP_TCP_PUT = 'P';  // inline block data put (emulated RDMA_WRITE)
constexpr char OP_TCP_GET = 'G';  // inline block data get (emulated server push)
constexpr char OP_DELETE = 'X';   // delete keys (extension: engine-driven eviction)
constexpr char OP_STATS = 'Q';    // server stats JSON over the wire (extension)
// Packed fast-path local ops (extension): same semantics as OP_W/OP_R but a
// flat binary layout the hot path can build/parse at memcpy speed — the
// flatbuffers LocalMetaRequest ops remain accepted for wire compatibility.
// Body: PackedLocalHdr, u64 offsets[n], NUL-separated key bytes (n keys).
constexpr char OP_W_FAST = 'w';
constexpr char OP_R_FAST = 'r';
// Shared-memory ring handshake (extension, same-host clients): body is the
// shm_open name of a client-created segment (csrc/core/shm_ring.h); after a
// FINISH response the client sends OP_W_FAST/OP_R_FAST/OP_SYNC through the
// ring instead of the socket.
constexpr char OP_SHM_SETUP = 'h';

#pragma pack(push, 1)
struct PackedLocalHdr {
    int32_t device;
    int32_t pid;
    uint64_t base_ptr;
    uint64_t base_offset;
    uint32_t block_size;
    uint32_t n_blocks;
    uint32_t flags;
    uint32_t rsvd;
    uint8_t ipc[64];
};
#pragma pack(pop)
static_assert(sizeof(PackedLocalHdr) == 104, "packed local header size");

// PackedLocalHdr.flags: defer the response until the copy completes (one
// round trip instead of request-ack + sync).
constexpr uint32_t kLocalFlagSyncResponse = 1;
// Store the (bf16) payload quantized to fp8 e4m3 with one scale per block —
// half the HBM per cached page; reads dequantize transparently (extension,
// csrc/gpu/gpu.hip quant kernels).
constexpr uint32_t kLocalFlagQuantFp8 = 2;

std::string op_name(char op);

// ---- status codes ---------------------------------------------------------
constexpr int INVALID_REQ = 400;
constexpr int FINISH = 200;
constexpr int TASK_ACCEPTED = 202;
constexpr int INTERNAL_ERROR = 500;
constexpr int KEY_NOT_FOUND = 404;
constexpr int RETRY = 408;
constexpr int SYSTEM_ERROR = 503;
constexpr int OUT_OF_MEMORY = 507;

constexpr size_t kProtocolBufferSize = 4u << 20;  // max body for one request

// Flow-control constants (same roles as the reference's WR constants,
// protocol.h:23-34; the TCP fabric uses them to bound in-flight bytes).
constexpr int kMaxWrBatch = 32;
constexpr int kMaxOutstandingWrites = 4096;

// ---- typed messages -------------------------------------------------------
struct RemoteBlockWire {  // matches flatbuffers struct RemoteBlock (16 B)
    uint32_t rkey;
    uint32_t pad_ = 0;
    uint64_t remote_addr;
};
static_assert(sizeof(RemoteBlockWire) == 16, "RemoteBlock wire size");

struct KeyOffset {
    std::string key;
    uint64_t offset;
};

struct LocalMetaMsg {
    int32_t device = 0;
    std::vector<uint8_t> ipc_handle;
    int32_t block_size = 0;
    std::vector<KeyOffset> blocks;
    uint64_t base_offset = 0;  // extension, field id 4
    int32_t pid = 0;           // extension, field id 5: client pid — enables a
    uint64_t base_ptr = 0;     // field id 6: same-process fast path (an IPC
                               // handle cannot be opened in its own process)
};

struct RemoteMetaMsg {
    std::vector<std::string> keys;
    int32_t block_size = 0;
    uint32_t rkey = 0;
    std::vector<uint64_t> remote_addrs;
    int8_t op = 0;
};

// Serializers produce a complete flatbuffer (root offset included).
std::vector<uint8_t> build_local_meta(const LocalMetaMsg& m);
bool parse_local_meta(const uint8_t* buf, size_t len, LocalMetaMsg* out);

std::vector<uint8_t> build_remote_meta(const RemoteMetaMsg& m);
bool parse_remote_meta(const uint8_t* buf, size_t len, RemoteMetaMsg* out);

std::vector<uint8_t> build_allocate_response(const std::vector<RemoteBlockWire>& blocks);
bool parse_allocate_response(const uint8_t* buf, size_t len, std::vector<RemoteBlockWire>* out);

std::vector<uint8_t> build_match_request(const std::vector<std::string>& keys);
bool parse_match_request(const uint8_t* buf, size_t len, std::vector<std::string>* out);

// Sentinel for duplicate-key allocations: the server returns rkey=0, addr=0
// for keys that already exist; clients skip writing those blocks (first write
// wins, matching the reference's dedup semantics).
inline bool is_fake_remote_block(uint32_t rkey, uint64_t addr) { return rkey == 0 && addr == 0; }

}  // namespace ifs
