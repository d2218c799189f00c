#include "protocol.h"

namespace ifs {

std::string op_name(char op) {
    switch (op) {
        case OP_R: return "local_read";
        case OP_W: return "local_write";
        case OP_SYNC: return "sync";
        case OP_RDMA_EXCHANGE: return "exchange";
        case OP_RDMA_ALLOCATE: return "allocate";
        case OP_RDMA_READ: return "rdma_read";
        case OP_RDMA_WRITE_COMMIT: return "write_commit";
        case OP_CHECK_EXIST: return "check_exist";
        case OP_GET_MATCH_LAST_IDX: return "match_last_index";
        case OP_TCP_PUT: return "tcp_put";
        case OP_TCP_GET: return "tcp_get";
        case OP_DELETE: return "delete";
        case OP_STATS: return "stats";
        case OP_W_FAST: return "local_write_fast";
        case OP_R_FAST: return "local_read_fast";
        case OP_SHM_SETUP: return "shm_setup";
        default: return "unknown";
    }
}

// --- LocalMetaRequest: device(0), ipc_handle(1), block_size(2), blocks(3),
//     base_offset(4 — extension). Block: key(0), offset(1).
std::vector<uint8_t> build_local_meta(const LocalMetaMsg& m) {
    wire::Builder b(512 + m.blocks.size() * 64);
    // Build Block subtables first (file-later objects are pushed first).
    std::vector<wire::Builder::Offset> block_offs;
    block_offs.reserve(m.blocks.size());
    for (auto it = m.blocks.rbegin(); it != m.blocks.rend(); ++it) {
        auto koff = b.create_string(it->key);
        b.start_table();
        b.add_offset(0, koff);
        b.add_scalar<uint64_t>(1, it->offset, 0);
        block_offs.push_back(b.end_table());
    }
    // block_offs built in reverse; restore order for the vector.
    std::vector<wire::Builder::Offset> ordered(block_offs.rbegin(), block_offs.rend());
    auto blocks_vec = b.create_offset_vector(ordered);
    auto ipc_vec = b.create_vector<uint8_t>(m.ipc_handle.data(), m.ipc_handle.size());
    b.start_table();
    b.add_scalar<int32_t>(0, m.device, 0);
    b.add_offset(1, ipc_vec);
    b.add_scalar<int32_t>(2, m.block_size, 0);
    b.add_offset(3, blocks_vec);
    b.add_scalar<uint64_t>(4, m.base_offset, 0);
    b.add_scalar<int32_t>(5, m.pid, 0);
    b.add_scalar<uint64_t>(6, m.base_ptr, 0);
    auto root = b.end_table();
    b.finish(root);
    return b.release();
}

bool parse_local_meta(const uint8_t* buf, size_t len, LocalMetaMsg* out) {
    wire::Reader r(buf, len);
    if (!r.ok()) return false;
    auto t = r.root();
    if (!t.valid()) return false;
    out->device = t.scalar<int32_t>(0, 0);
    out->ipc_handle = t.scalar_vector<uint8_t>(1);
    out->block_size = t.scalar<int32_t>(2, 0);
    out->base_offset = t.scalar<uint64_t>(4, 0);
    out->pid = t.scalar<int32_t>(5, 0);
    out->base_ptr = t.scalar<uint64_t>(6, 0);
    out->blocks.clear();
    size_t n = t.vec_len(3);
    out->blocks.reserve(n);
    for (size_t i = 0; i < n; i++) {
        auto bt = t.table_at_vec(3, i);
        if (!bt.valid()) return false;
        out->blocks.push_back({bt.string_field(0), bt.scalar<uint64_t>(1, 0)});
    }
    return true;
}

// --- RemoteMetaRequest: keys(0), block_size(1), rkey(2), remote_addrs(3), op(4)
std::vector<uint8_t> build_remote_meta(const RemoteMetaMsg& m) {
    wire::Builder b(256 + m.keys.size() * 48 + m.remote_addrs.size() * 8);
    std::vector<wire::Builder::Offset> key_offs;
    key_offs.reserve(m.keys.size());
    for (auto it = m.keys.rbegin(); it != m.keys.rend(); ++it) key_offs.push_back(b.create_string(*it));
    std::vector<wire::Builder::Offset> ordered(key_offs.rbegin(), key_offs.rend());
    auto keys_vec = b.create_offset_vector(ordered);
    auto addrs_vec = b.create_vector<uint64_t>(m.remote_addrs.data(), m.remote_addrs.size());
    b.start_table();
    b.add_offset(0, keys_vec);
    b.add_scalar<int32_t>(1, m.block_size, 0);
    b.add_scalar<uint32_t>(2, m.rkey, 0);
    b.add_offset(3, addrs_vec);
    b.add_scalar<int8_t>(4, m.op, 0);
    auto root = b.end_table();
    b.finish(root);
    return b.release();
}

bool parse_remote_meta(const uint8_t* buf, size_t len, RemoteMetaMsg* out) {
    wire::Reader r(buf, len);
    if (!r.ok()) return false;
    auto t = r.root();
    if (!t.valid()) return false;
    out->keys = t.string_vector(0);
    out->block_size = t.scalar<int32_t>(1, 0);
    out->rkey = t.scalar<uint32_t>(2, 0);
    out->remote_addrs = t.scalar_vector<uint64_t>(3);
    out->op = t.scalar<int8_t>(4, 0);
    return true;
}

// --- RdmaAllocateResponse: blocks(0) = vector of RemoteBlock struct (16 B)
std::vector<uint8_t> build_allocate_response(const std::vector<RemoteBlockWire>& blocks) {
    wire::Builder b(64 + blocks.size() * 16);
    auto vec = b.create_struct_vector(blocks.data(), blocks.size(), sizeof(RemoteBlockWire), 8);
    b.start_table();
    b.add_offset(0, vec);
    auto root = b.end_table();
    b.finish(root);
    return b.release();
}

bool parse_allocate_response(const uint8_t* buf, size_t len, std::vector<RemoteBlockWire>* out) {
    wire::Reader r(buf, len);
    if (!r.ok()) return false;
    auto t = r.root();
    if (!t.valid()) return false;
    *out = t.scalar_vector<RemoteBlockWire>(0);
    return true;
}

// --- GetMatchLastIndexRequest: keys(0)
std::vector<uint8_t> build_match_request(const std::vector<std::string>& keys) {
    wire::Builder b(64 + keys.size() * 48);
    std::vector<wire::Builder::Offset> key_offs;
    key_offs.reserve(keys.size());
    for (auto it = keys.rbegin(); it != keys.rend(); ++it) key_offs.push_back(b.create_string(*it));
    std::vector<wire::Builder::Offset> ordered(key_offs.rbegin(), key_offs.rend());
    auto keys_vec = b.create_offset_vector(ordered);
    b.start_table();
    b.add_offset(0, keys_vec);
    auto root = b.end_table();
    b.finish(root);
    return b.release();
}

bool parse_match_request(const uint8_t* buf, size_t len, std::vector<std::string>* out) {
    wire::Reader r(buf, len);
    if (!r.ok()) return false;
    auto t = r.root();
    if (!t.valid()) return false;
    *out = t.string_vector(0);
    return true;
}

}  // namespace ifs
