// Python bindings for infinistore-amd (module: infinistore_amd._native).
//
// Mirrors the reference's pybind surface (/root/reference/src/pybind.cpp:36-210:
// Connection, ClientConfig/ServerConfig, register_server, purge_kv_map,
// get_kvmap_len, log) with two deliberate changes:
//  * The server runs on its own C++ thread (start_server/stop_server) instead
//    of borrowing uvloop's uv_loop_t* through a PyCapsule.
//  * allocate_rdma returns a list of (rkey, remote_addr) tuples and w_rdma
//    accepts the same — no numpy dtype marshalling required.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <memory>

#include "client/client.h"
#include "core/log.h"
#include "fabric/wr_flow.h"
#include "core/protocol.h"
#include "core/mempool.h"
#include "gpu/gpu.h"
#include "server/server.h"

namespace py = pybind11;
using namespace ifs;

namespace {

// Hold a Python callback so it can be copied/destroyed on C++ worker threads:
// the shared_ptr deleter re-acquires the GIL for the final decref.
std::shared_ptr<py::function> hold_callback(py::function cb) {
    return std::shared_ptr<py::function>(new py::function(std::move(cb)), [](py::function* p) {
        py::gil_scoped_acquire acq;
        delete p;
    });
}

std::unique_ptr<Server> g_server;

struct ServerConfigPy {
    int manage_port = 0;
    int service_port = 0;
    std::string log_level = "warning";
    std::string dev_name = "";
    int ib_port = 1;
    std::string link_type = "Ethernet";
    int prealloc_size = 16;           // GB per shard
    int minimal_allocate_size = 64;   // KB
    int num_stream = 4;
    bool auto_increase = false;
    std::vector<int> devices;         // GPU ordinals to shard over; empty=auto
    bool cpu_only = false;            // force CPU pool even if GPUs exist
    int cpu_shards = 1;               // CPU-mode shard count
    bool auto_evict = false;          // LRU-evict on allocation failure
    int ttl_seconds = 0;              // key time-to-live (0 = forever)
    int io_threads = 3;               // worker IO loops (0 = single loop)
    int extend_size = 10;             // GB per auto-extend arena
};

bool start_server(const ServerConfigPy& cfg) {
    if (g_server && g_server->running()) {
        ERROR("server already running");
        return false;
    }
    ServerOptions opt;
    opt.service_port = cfg.service_port;
    opt.prealloc_bytes = static_cast<size_t>(cfg.prealloc_size) << 30;
    opt.block_granule = static_cast<size_t>(cfg.minimal_allocate_size) << 10;
    opt.auto_extend = cfg.auto_increase;
    opt.n_streams = cfg.num_stream > 0 ? cfg.num_stream : 4;
    opt.cpu_shards = cfg.cpu_shards;
    opt.log_level = cfg.log_level;
    opt.dev_name = cfg.dev_name;
    opt.ib_port = cfg.ib_port;
    opt.link_type = cfg.link_type;
    opt.auto_evict = cfg.auto_evict;
    opt.ttl_seconds = cfg.ttl_seconds;
    opt.io_threads = cfg.io_threads;
    opt.extend_bytes = static_cast<size_t>(cfg.extend_size) << 30;
    if (!cfg.cpu_only && gpu::available()) {
        if (!cfg.devices.empty()) {
            opt.devices = cfg.devices;
        } else {
            opt.devices.resize(static_cast<size_t>(gpu::device_count()));
            for (int i = 0; i < gpu::device_count(); i++) opt.devices[static_cast<size_t>(i)] = i;
        }
    }
    g_server.reset(new Server(opt));
    return g_server->start();
}

void stop_server() {
    if (g_server) {
        g_server->stop();
        g_server.reset();
    }
}

size_t purge_kv_map_py() { return g_server ? g_server->purge() : 0; }
size_t get_kvmap_len_py() { return g_server ? g_server->kvmap_len() : 0; }
std::string server_stats_py() { return g_server ? g_server->stats_json() : "{}"; }
std::pair<size_t, size_t> server_compact_py() {
    return g_server ? g_server->compact() : std::make_pair<size_t, size_t>(0, 0);
}

// GPU fingerprint helper: hash n blocks of a device tensor in one kernel
// launch (block i at base + offsets[i], each `block_size` bytes).
std::vector<uint64_t> hash_blocks_py(uintptr_t base, std::vector<uint64_t> offsets,
                                     size_t block_size, int device) {
    size_t n = offsets.size();
    std::vector<uint64_t> out(n, 0);
    if (!gpu::available() || n == 0) return out;
    std::vector<uint64_t> ptrs(n);
    for (size_t i = 0; i < n; i++) ptrs[i] = base + offsets[i];
    gpu::set_device(device);
    auto* d_ptrs = static_cast<uint64_t*>(gpu::alloc_device(device, n * 8));
    auto* d_out = static_cast<uint64_t*>(gpu::alloc_device(device, n * 8));
    if (!d_ptrs || !d_out) return out;
    gpu::Stream s = gpu::stream_create(device);
    gpu::memcpy_h2d(d_ptrs, ptrs.data(), n * 8);
    gpu::launch_hash_blocks(device, s, d_ptrs, static_cast<int>(n), block_size, d_out);
    gpu::stream_sync(s);
    gpu::memcpy_d2h(out.data(), d_out, n * 8);
    gpu::stream_destroy(s);
    gpu::free_device(d_ptrs);
    gpu::free_device(d_out);
    return out;
}

}  // namespace

PYBIND11_MODULE(_native, m) {
    m.doc() = "infinistore-amd native core (MI355X / ROCm)";

    // ---- configs ----
    py::class_<ClientConfigC>(m, "ClientConfig")
        .def(py::init<>())
        .def_readwrite("host_addr", &ClientConfigC::host_addr)
        .def_readwrite("service_port", &ClientConfigC::service_port)
        .def_readwrite("connection_type", &ClientConfigC::connection_type)
        .def_readwrite("dev_name", &ClientConfigC::dev_name)
        .def_readwrite("ib_port", &ClientConfigC::ib_port)
        .def_readwrite("link_type", &ClientConfigC::link_type)
        .def_readwrite("log_level", &ClientConfigC::log_level);

    py::class_<ServerConfigPy>(m, "ServerConfig")
        .def(py::init<>())
        .def_readwrite("manage_port", &ServerConfigPy::manage_port)
        .def_readwrite("service_port", &ServerConfigPy::service_port)
        .def_readwrite("log_level", &ServerConfigPy::log_level)
        .def_readwrite("dev_name", &ServerConfigPy::dev_name)
        .def_readwrite("ib_port", &ServerConfigPy::ib_port)
        .def_readwrite("link_type", &ServerConfigPy::link_type)
        .def_readwrite("prealloc_size", &ServerConfigPy::prealloc_size)
        .def_readwrite("minimal_allocate_size", &ServerConfigPy::minimal_allocate_size)
        .def_readwrite("num_stream", &ServerConfigPy::num_stream)
        .def_readwrite("auto_increase", &ServerConfigPy::auto_increase)
        .def_readwrite("devices", &ServerConfigPy::devices)
        .def_readwrite("cpu_only", &ServerConfigPy::cpu_only)
        .def_readwrite("cpu_shards", &ServerConfigPy::cpu_shards)
        .def_readwrite("auto_evict", &ServerConfigPy::auto_evict)
        .def_readwrite("ttl_seconds", &ServerConfigPy::ttl_seconds)
        .def_readwrite("io_threads", &ServerConfigPy::io_threads)
        .def_readwrite("extend_size", &ServerConfigPy::extend_size);

    // ---- client connection ----
    py::class_<ClientConn>(m, "Connection")
        .def(py::init<>())
        .def("init_connection", &ClientConn::init_connection,
             py::call_guard<py::gil_scoped_release>())
        .def("setup_rdma", &ClientConn::setup_rdma, py::call_guard<py::gil_scoped_release>())
        .def("close_conn", &ClientConn::close_conn, py::call_guard<py::gil_scoped_release>())
        .def(
            "rw_local",
            [](ClientConn& c, const std::string& op,
               const std::vector<std::pair<std::string, uint64_t>>& blocks, int block_size,
               uintptr_t ptr, int device) {
                py::gil_scoped_release rel;
                return c.rw_local(op.empty() ? 'W' : op[0], blocks, block_size, ptr, device);
            })
        .def(
            "rw_local_fast",
            [](ClientConn& c, const std::string& op, py::bytes keys_blob, py::bytes offsets,
               size_t n, int block_size, uintptr_t ptr, int device, bool sync_response) {
                char* kb;
                Py_ssize_t kb_len;
                PyBytes_AsStringAndSize(keys_blob.ptr(), &kb, &kb_len);
                char* ob;
                Py_ssize_t ob_len;
                PyBytes_AsStringAndSize(offsets.ptr(), &ob, &ob_len);
                if (static_cast<size_t>(ob_len) != n * 8)
                    throw std::runtime_error("offsets must be n uint64");
                py::gil_scoped_release rel;
                return c.rw_local_packed(op.empty() ? 'W' : op[0], kb,
                                         static_cast<size_t>(kb_len),
                                         reinterpret_cast<const uint64_t*>(ob), n, block_size,
                                         ptr, device, sync_response);
            },
            py::arg("op"), py::arg("keys_blob"), py::arg("offsets"), py::arg("n"),
            py::arg("block_size"), py::arg("ptr"), py::arg("device"),
            py::arg("sync_response") = false)
        .def(
            "rw_local_keys",
            [](ClientConn& c, const std::string& op, py::list keys, py::buffer offsets,
               uint64_t element_size, int block_size, uintptr_t ptr, int device,
               bool sync_response, uint32_t extra_flags) {
                // Build the NUL-joined key blob and byte offsets in C++ —
                // no Python-side join/numpy work on the hot path.
                py::buffer_info ob = offsets.request();
                size_t n = static_cast<size_t>(py::len(keys));
                if (ob.itemsize != 8 || static_cast<size_t>(ob.size) < n)
                    throw std::runtime_error("offsets must be uint64[n]");
                const uint64_t* offs = static_cast<const uint64_t*>(ob.ptr);
                std::string blob;
                blob.reserve(n * 48);
                std::vector<uint64_t> byte_offs(n);
                for (size_t i = 0; i < n; i++) {
                    Py_ssize_t klen = 0;
                    const char* ks =
                        PyUnicode_AsUTF8AndSize(keys[i].ptr(), &klen);
                    if (!ks) throw std::runtime_error("keys must be str");
                    if (i) blob.push_back('\0');
                    blob.append(ks, static_cast<size_t>(klen));
                    byte_offs[i] = offs[i] * element_size;
                }
                py::gil_scoped_release rel;
                return c.rw_local_packed(op.empty() ? 'W' : op[0], blob.data(), blob.size(),
                                         byte_offs.data(), n, block_size, ptr, device,
                                         sync_response, nullptr, extra_flags);
            },
            py::arg("op"), py::arg("keys"), py::arg("offsets"), py::arg("element_size"),
            py::arg("block_size"), py::arg("ptr"), py::arg("device"),
            py::arg("sync_response") = false, py::arg("extra_flags") = 0)
        .def(
            "rw_local_blob",
            [](ClientConn& c, const std::string& op, py::buffer blob, py::buffer offsets,
               uint64_t element_size, int block_size, uintptr_t ptr, int device,
               bool sync_response, uint32_t extra_flags) {
                // Zero-pack fast path: keys as one NUL-separated bytes blob
                // (engines cache the serialized page-key chain; re-joining
                // 2k Python strings costs ~30 µs per request otherwise).
                py::buffer_info bb = blob.request();
                py::buffer_info ob = offsets.request();
                if (ob.itemsize != 8) throw std::runtime_error("offsets must be uint64[n]");
                size_t n = static_cast<size_t>(ob.size);
                const uint64_t* offs = static_cast<const uint64_t*>(ob.ptr);
                std::vector<uint64_t> byte_offs(n);
                for (size_t i = 0; i < n; i++) byte_offs[i] = offs[i] * element_size;
                const char* bp = static_cast<const char*>(bb.ptr);
                size_t blen = static_cast<size_t>(bb.size) * bb.itemsize;
                py::gil_scoped_release rel;
                return c.rw_local_packed(op.empty() ? 'W' : op[0], bp, blen, byte_offs.data(),
                                         n, block_size, ptr, device, sync_response, nullptr,
                                         extra_flags);
            },
            py::arg("op"), py::arg("blob"), py::arg("offsets"), py::arg("element_size"),
            py::arg("block_size"), py::arg("ptr"), py::arg("device"),
            py::arg("sync_response") = false, py::arg("extra_flags") = 0)
        .def(
            "rw_local_blob_async",
            [](ClientConn& c, const std::string& op, py::buffer blob, py::buffer offsets,
               uint64_t element_size, int block_size, uintptr_t ptr, int device) {
                // Ticketed form: push the (sync-response) op and return a
                // ticket to pass to wait_local_ticket — the engine overlaps
                // its own work with the copy. Ticket 0 = already complete
                // (op fell back to the blocking socket path).
                py::buffer_info bb = blob.request();
                py::buffer_info ob = offsets.request();
                if (ob.itemsize != 8) throw std::runtime_error("offsets must be uint64[n]");
                size_t n = static_cast<size_t>(ob.size);
                const uint64_t* offs = static_cast<const uint64_t*>(ob.ptr);
                std::vector<uint64_t> byte_offs(n);
                for (size_t i = 0; i < n; i++) byte_offs[i] = offs[i] * element_size;
                const char* bp = static_cast<const char*>(bb.ptr);
                size_t blen = static_cast<size_t>(bb.size) * bb.itemsize;
                uint64_t ticket = 0;
                int ret;
                {
                    py::gil_scoped_release rel;
                    ret = c.rw_local_packed(op.empty() ? 'R' : op[0], bp, blen,
                                            byte_offs.data(), n, block_size, ptr, device,
                                            /*sync_response=*/true, &ticket);
                }
                return py::make_tuple(ret, ticket);
            },
            py::arg("op"), py::arg("blob"), py::arg("offsets"), py::arg("element_size"),
            py::arg("block_size"), py::arg("ptr"), py::arg("device"))
        .def("wait_local_ticket", &ClientConn::wait_local_ticket,
             py::call_guard<py::gil_scoped_release>())
        .def("sync_local", &ClientConn::sync_local, py::call_guard<py::gil_scoped_release>())
        .def("register_mr", &ClientConn::register_mr, py::call_guard<py::gil_scoped_release>())
        .def(
            "allocate_rdma",
            [](ClientConn& c, const std::vector<std::string>& keys, int block_size) {
                std::vector<RemoteBlockOut> res;
                {
                    py::gil_scoped_release rel;
                    res = c.allocate_rdma(keys, block_size);
                }
                py::list out;
                for (auto& b : res) out.append(py::make_tuple(b.rkey, b.remote_addr));
                return out;
            })
        .def(
            "allocate_rdma_async",
            [](ClientConn& c, const std::vector<std::string>& keys, int block_size,
               py::function cb) {
                auto cbp = hold_callback(std::move(cb));
                auto cb_cpp = [cbp](std::vector<RemoteBlockOut> res) {
                    py::gil_scoped_acquire acq;
                    py::list out;
                    for (auto& b : res) out.append(py::make_tuple(b.rkey, b.remote_addr));
                    (*cbp)(out);
                };
                py::gil_scoped_release rel;
                return c.allocate_rdma_async(keys, block_size, std::move(cb_cpp));
            })
        .def(
            "w_rdma",
            [](ClientConn& c, const std::vector<uint64_t>& offsets, int block_size,
               const std::vector<std::pair<uint32_t, uint64_t>>& remote_blocks, uintptr_t ptr) {
                std::vector<RemoteBlockOut> blks;
                blks.reserve(remote_blocks.size());
                for (auto& rb : remote_blocks) blks.push_back({rb.first, rb.second});
                py::gil_scoped_release rel;
                return c.w_rdma(offsets.data(), offsets.size(), block_size, blks.data(),
                                blks.size(), ptr);
            })
        .def(
            "w_rdma_async",
            [](ClientConn& c, const std::vector<uint64_t>& offsets, int block_size,
               const std::vector<std::pair<uint32_t, uint64_t>>& remote_blocks, uintptr_t ptr,
               py::function cb) {
                std::vector<RemoteBlockOut> blks;
                blks.reserve(remote_blocks.size());
                for (auto& rb : remote_blocks) blks.push_back({rb.first, rb.second});
                auto cbp = hold_callback(std::move(cb));
                auto cb_cpp = [cbp]() {
                    py::gil_scoped_acquire acq;
                    (*cbp)();
                };
                py::gil_scoped_release rel;
                return c.w_rdma_async(offsets.data(), offsets.size(), block_size, blks.data(),
                                      blks.size(), ptr, std::move(cb_cpp));
            })
        .def(
            "r_rdma",
            [](ClientConn& c, const std::vector<std::pair<std::string, uint64_t>>& blocks,
               int block_size, uintptr_t ptr) {
                py::gil_scoped_release rel;
                return c.r_rdma(blocks, block_size, ptr);
            })
        .def(
            "r_rdma_async",
            [](ClientConn& c, const std::vector<std::pair<std::string, uint64_t>>& blocks,
               int block_size, uintptr_t ptr, py::function cb) {
                auto cbp = hold_callback(std::move(cb));
                auto cb_cpp = [cbp]() {
                    py::gil_scoped_acquire acq;
                    (*cbp)();
                };
                py::gil_scoped_release rel;
                return c.r_rdma_async(blocks, block_size, ptr, std::move(cb_cpp));
            })
        .def("sync_rdma", &ClientConn::sync_rdma, py::call_guard<py::gil_scoped_release>())
        .def("check_exist", &ClientConn::check_exist, py::call_guard<py::gil_scoped_release>())
        .def("get_match_last_index", &ClientConn::get_match_last_index,
             py::call_guard<py::gil_scoped_release>())
        .def("delete_keys", &ClientConn::delete_keys, py::call_guard<py::gil_scoped_release>())
        .def("get_stats", &ClientConn::get_stats, py::call_guard<py::gil_scoped_release>())
        .def("shm_active", &ClientConn::shm_active)
        .def("using_verbs", &ClientConn::using_verbs);

    // ---- server ----
    m.def("start_server", &start_server, py::call_guard<py::gil_scoped_release>());
    m.def("stop_server", &stop_server, py::call_guard<py::gil_scoped_release>());
    m.def("purge_kv_map", &purge_kv_map_py);
    m.def("get_kvmap_len", &get_kvmap_len_py);
    m.def("server_stats", &server_stats_py);
    m.def("server_compact", &server_compact_py, py::call_guard<py::gil_scoped_release>());
    m.def("server_snapshot", [](const std::string& path) {
        std::pair<size_t, size_t> r{0, 0};
        bool ok;
        {
            py::gil_scoped_release rel;
            ok = g_server && g_server->snapshot(path, &r);
        }
        return py::make_tuple(ok, r.first, r.second);
    });
    m.def("server_restore", [](const std::string& path) {
        std::pair<size_t, size_t> r{0, 0};
        bool ok;
        {
            py::gil_scoped_release rel;
            ok = g_server && g_server->restore(path, &r);
        }
        return py::make_tuple(ok, r.first, r.second);
    });

    // ---- logging ----
    m.def("set_log_level", [](const std::string& lvl) { set_log_level(lvl.c_str()); });
    m.def("log_msg", [](const std::string& lvl, const std::string& msg) {
        LogLevel l = LogLevel::kInfo;
        if (lvl == "debug") l = LogLevel::kDebug;
        else if (lvl == "warning" || lvl == "warn") l = LogLevel::kWarn;
        else if (lvl == "error") l = LogLevel::kError;
        log_printf(l, "python", 0, "%s", msg.c_str());
    });

    // ---- GPU info / kernels ----
    m.def("gpu_available", [] { return gpu::available(); });
    m.def("gpu_count", [] { return gpu::device_count(); });
    m.def("hash_blocks", &hash_blocks_py, py::call_guard<py::gil_scoped_release>());

    // ---- debug/test surface (wire format + mempool) ----
    m.def("_dbg_build_local_meta", [](int device, py::bytes ipc, int block_size,
                                      const std::vector<std::pair<std::string, uint64_t>>& blocks,
                                      uint64_t base_offset) {
        LocalMetaMsg msg;
        msg.device = device;
        std::string s = ipc;
        msg.ipc_handle.assign(s.begin(), s.end());
        msg.block_size = block_size;
        msg.base_offset = base_offset;
        for (auto& b : blocks) msg.blocks.push_back({b.first, b.second});
        auto v = build_local_meta(msg);
        return py::bytes(reinterpret_cast<const char*>(v.data()), v.size());
    });
    // Export a device pointer's IPC handle + allocation offset: lets tests
    // hand-craft reference-framed LocalMetaRequest transcripts (the 'W'/'R'
    // flatbuffers ops) without going through this client library.
    m.def("_dbg_ipc_export", [](uintptr_t ptr) {
        ifs::gpu::IpcHandle h;
        uint64_t off = 0;
        if (!ifs::gpu::ipc_export(reinterpret_cast<void*>(ptr), &h, &off))
            throw std::runtime_error("ipc_export failed");
        return py::make_tuple(
            py::bytes(reinterpret_cast<const char*>(h.bytes), sizeof(h.bytes)), off);
    });

    m.def("_dbg_parse_local_meta", [](py::bytes data) {
        std::string s = data;
        LocalMetaMsg msg;
        if (!parse_local_meta(reinterpret_cast<const uint8_t*>(s.data()), s.size(), &msg))
            throw std::runtime_error("parse failed");
        py::list blocks;
        for (auto& b : msg.blocks) blocks.append(py::make_tuple(b.key, b.offset));
        return py::make_tuple(msg.device, py::bytes(reinterpret_cast<char*>(msg.ipc_handle.data()),
                                                    msg.ipc_handle.size()),
                              msg.block_size, blocks, msg.base_offset);
    });
    m.def("_dbg_build_remote_meta",
          [](const std::vector<std::string>& keys, int block_size, uint32_t rkey,
             const std::vector<uint64_t>& addrs, int op) {
              RemoteMetaMsg msg{keys, block_size, rkey, addrs, static_cast<int8_t>(op)};
              auto v = build_remote_meta(msg);
              return py::bytes(reinterpret_cast<const char*>(v.data()), v.size());
          });
    m.def("_dbg_parse_remote_meta", [](py::bytes data) {
        std::string s = data;
        RemoteMetaMsg msg;
        if (!parse_remote_meta(reinterpret_cast<const uint8_t*>(s.data()), s.size(), &msg))
            throw std::runtime_error("parse failed");
        return py::make_tuple(msg.keys, msg.block_size, msg.rkey, msg.remote_addrs,
                              static_cast<int>(msg.op));
    });
    m.def("_dbg_build_alloc_resp", [](const std::vector<std::pair<uint32_t, uint64_t>>& blocks) {
        std::vector<RemoteBlockWire> w;
        for (auto& b : blocks) w.push_back({b.first, 0, b.second});
        auto v = build_allocate_response(w);
        return py::bytes(reinterpret_cast<const char*>(v.data()), v.size());
    });
    m.def("_dbg_parse_alloc_resp", [](py::bytes data) {
        std::string s = data;
        std::vector<RemoteBlockWire> w;
        if (!parse_allocate_response(reinterpret_cast<const uint8_t*>(s.data()), s.size(), &w))
            throw std::runtime_error("parse failed");
        py::list out;
        for (auto& b : w) out.append(py::make_tuple(b.rkey, b.remote_addr));
        return out;
    });
    m.def("_dbg_build_match_req", [](const std::vector<std::string>& keys) {
        auto v = build_match_request(keys);
        return py::bytes(reinterpret_cast<const char*>(v.data()), v.size());
    });
    m.def("_dbg_parse_match_req", [](py::bytes data) {
        std::string s = data;
        std::vector<std::string> keys;
        if (!parse_match_request(reinterpret_cast<const uint8_t*>(s.data()), s.size(), &keys))
            throw std::runtime_error("parse failed");
        return keys;
    });

    // WR flow-control simulation (chain split + outstanding cap + overflow).
    // shm ring framing/wrap test harness: push `msgs` through a tiny ring
    // with an interleaved consumer, return the bodies read back in order.
    m.def("_dbg_shmring_echo", [](const std::vector<py::bytes>& msgs, uint32_t cap,
                                  int drain_every) {
        std::vector<uint8_t> mem(ifs::shmring::Ring::footprint(cap));
        auto* ring = reinterpret_cast<ifs::shmring::Ring*>(mem.data());
        new (&ring->head) std::atomic<uint64_t>(0);
        new (&ring->tail) std::atomic<uint64_t>(0);
        ring->cap = cap;
        std::vector<py::bytes> out;
        uint64_t seq = 0;
        auto drain = [&] {
            for (;;) {
                uint32_t len = 0;
                uint64_t skip = 0;
                const uint8_t* rec = ring->peek(&len, &skip);
                if (!rec) return;
                if (len) {
                    ifs::shmring::RecHdr h;
                    memcpy(&h, rec, sizeof(h));
                    if (h.seq != ++seq) throw std::runtime_error("seq out of order");
                    out.emplace_back(reinterpret_cast<const char*>(rec + sizeof(h)),
                                     h.body_len);
                }
                ring->consume(skip);
            }
        };
        int pushed = 0;
        for (auto& m2 : msgs) {
            std::string body = m2;
            uint32_t need = ifs::shmring::rec_len(body.size());
            uint64_t adv = 0;
            uint8_t* dst = ring->claim(need, &adv);
            for (long spin = 0; !dst; spin++) {  // full: drain like the consumer would
                drain();
                dst = ring->claim(need, &adv);
                if (spin > 1000000) throw std::runtime_error("ring stuck");
            }
            ifs::shmring::RecHdr h{};
            h.len = need;
            h.op = 'T';
            h.body_len = static_cast<uint32_t>(body.size());
            h.seq = static_cast<uint64_t>(pushed + 1);
            memcpy(dst, &h, sizeof(h));
            memcpy(dst + sizeof(h), body.data(), body.size());
            ring->publish(adv);
            if (++pushed % std::max(1, drain_every) == 0) drain();
        }
        drain();
        return out;
    });
    m.def("_dbg_wrflow_sim", [](int n_wrs, int batch, int cap, int complete_after) {
        std::deque<size_t> inflight_chains;
        std::vector<size_t> posted_sizes;
        long peak = 0;
        WrFlow* flow_ptr = nullptr;
        WrFlow flow(
            [&](const WrChain& ch) {
                posted_sizes.push_back(ch.wrs.size());
                inflight_chains.push_back(ch.wrs.size());
                if (flow_ptr && flow_ptr->outstanding() > peak) peak = flow_ptr->outstanding();
                return true;
            },
            batch, cap);
        flow_ptr = &flow;
        (void)complete_after;
        std::vector<WrDesc> wrs(static_cast<size_t>(n_wrs), WrDesc{0, 0, 0, 0, 0});
        flow.submit(std::move(wrs), true, 42, 7);
        size_t parked_peak = flow.parked();
        // Deliver completions FIFO; draining may post more chains.
        while (!inflight_chains.empty()) {
            size_t n = inflight_chains.front();
            inflight_chains.pop_front();
            flow.on_chain_complete(n);
        }
        return py::make_tuple(posted_sizes, peak, flow.outstanding(), parked_peak,
                              flow.parked());
    });

    // Mempool (host-backed) for allocator unit tests.
    py::class_<MemoryPool>(m, "_TestPool")
        .def(py::init([](size_t size, size_t block_size) {
                 void* base = malloc(size);
                 return new MemoryPool(base, size, block_size, 0);
             }),
             py::return_value_policy::take_ownership)
        .def("allocate",
             [](MemoryPool& p, size_t size) { return reinterpret_cast<uintptr_t>(p.allocate(size)); })
        .def("deallocate",
             [](MemoryPool& p, uintptr_t ptr, size_t size) {
                 return p.deallocate(reinterpret_cast<void*>(ptr), size);
             })
        .def("used_blocks", &MemoryPool::used_blocks)
        .def("total_blocks", &MemoryPool::total_blocks)
        .def("base", [](MemoryPool& p) { return reinterpret_cast<uintptr_t>(p.base()); });
}
