// Shared-memory request ring for same-host clients.
//
// The local-GPU path moves payloads with a batched HIP kernel, so the only
// thing left on the socket is the ~100 KB packed request and a 4-byte
// status — yet the UDS round trip plus the libuv dispatch hop costs tens of
// microseconds per request. This transport replaces the socket for the hot
// packed ops (OP_W_FAST / OP_R_FAST / OP_SYNC) with two SPSC byte rings in
// a client-created shm segment: the client pushes request records and
// spin-waits for fixed-size response records; the server polls the request
// ring on a dedicated thread and completion threads write responses
// directly (no loop-thread hop). Everything else — and any request that
// does not fit — stays on the socket.
//
// The reference has no equivalent (its local path rides the same TCP
// socket as everything else, /root/reference/src/infinistore.cpp:1122);
// this is an MI355X-native addition: at HBM3E speed the 128 KB-block
// gather kernel finishes 2048 blocks in ~90 µs, so host-side transport is
// the dominant cost and is worth designing out.
//
// Concurrency contract:
//   * request ring:  single producer (client, under its io mutex), single
//     consumer (server poller thread).
//   * response ring: single consumer (client); multiple server-side writers
//     (poller + per-shard completion threads) serialize on a server-local
//     mutex, so cross-process it still behaves as SPSC.
//   * head/tail are std::atomic<uint64_t> in shared memory (lock-free on
//     x86-64); release/acquire pairs order the record bytes.
#pragma once

#include <fcntl.h>
#include <sys/mman.h>
#include <unistd.h>

#include <atomic>
#include <cstdint>
#include <cstring>
#include <string>

namespace ifs {
namespace shmring {

constexpr uint64_t kMagic = 0x49465352494e4731ull;  // "IFSRING1"
constexpr uint32_t kVersion = 1;
constexpr uint32_t kWrapMarker = 0xffffffffu;

// Record header (8-byte aligned; len includes the header and padding,
// body_len is the exact payload size — the packed-op parsers derive the key
// blob from it, so padding must not leak into the body).
struct RecHdr {
    uint32_t len;
    uint8_t op;
    uint8_t _pad[3];
    uint32_t body_len;
    uint32_t t_push_us;  // producer CLOCK_MONOTONIC µs (mod 2^32): same clock
                         // domain on one host, used for queue-delay tracing
    uint64_t seq;
};

inline uint32_t mono_us() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return static_cast<uint32_t>(ts.tv_sec * 1000000ull + ts.tv_nsec / 1000);
}
static_assert(sizeof(RecHdr) == 24, "RecHdr layout");

inline uint32_t rec_len(size_t body) {
    return static_cast<uint32_t>((sizeof(RecHdr) + body + 7) & ~size_t(7));
}

// Fixed-size response record: status only (payload-carrying ops stay on the
// socket).
struct RespRec {
    RecHdr h;  // op = 0
    int32_t status;
    uint32_t _pad;
};
static_assert(sizeof(RespRec) == 32, "RespRec layout");

// One direction. `data[cap]` follows the struct; cap is a multiple of 8 and
// records are 8-byte aligned, so a record never straddles the wrap point
// (a kWrapMarker len, or exact exhaustion, sends the reader back to 0).
struct Ring {
    std::atomic<uint64_t> head;  // bytes ever written (producer)
    std::atomic<uint64_t> tail;  // bytes ever consumed (consumer)
    uint32_t cap;
    uint32_t _pad;
    uint8_t data[];

    static size_t footprint(uint32_t cap) { return sizeof(Ring) + cap; }

    // ---- producer ----
    // Contiguous space for `need` bytes at the current head, or nullptr.
    // On success *adv is the head advance (need, plus any wrap skip).
    uint8_t* claim(uint32_t need, uint64_t* adv) {
        uint64_t h = head.load(std::memory_order_relaxed);
        uint64_t t = tail.load(std::memory_order_acquire);
        uint64_t free_b = cap - (h - t);
        uint32_t pos = static_cast<uint32_t>(h % cap);
        uint32_t until_end = cap - pos;
        if (need <= until_end) {
            if (free_b < need) return nullptr;
            *adv = need;
            return data + pos;
        }
        // Wrap: skip the tail of the buffer, start at 0.
        if (free_b < until_end + need) {
            // Not enough for skip+record in one go. If the tail region is
            // already free, burn it NOW (publish the skip alone) so the next
            // claim starts at offset 0 — otherwise a record with
            // until_end + need > cap could never be placed even on an
            // otherwise-empty ring.
            if (free_b >= until_end) {
                if (until_end >= 4) {
                    uint32_t m = kWrapMarker;
                    memcpy(data + pos, &m, 4);
                }
                publish(until_end);
            }
            return nullptr;  // caller retries after the consumer drains
        }
        if (until_end >= 4) {
            uint32_t m = kWrapMarker;
            memcpy(data + pos, &m, 4);
        }
        *adv = until_end + need;
        return data;
    }
    void publish(uint64_t adv) {
        head.store(head.load(std::memory_order_relaxed) + adv, std::memory_order_release);
    }

    // ---- consumer ----
    // Peek the next record; nullptr if empty. *len is the record's total
    // length. Caller must consume(*skip) after copying out.
    //
    // A correct producer publishes whole records with one release store of
    // head, so once `h - t >= 4` the length field is fully visible and any
    // invalid value is PROVABLY corrupt (hostile or broken peer), never a
    // torn write. In particular a record that would straddle the ring's end
    // (l > until_end) can only come from a peer writing the header by hand —
    // the producer's claim() always wraps first — and following it would
    // read past the mapped segment. On corruption we set *corrupt and return
    // nullptr; the caller must stop consuming this ring for good.
    const uint8_t* peek(uint32_t* len, uint64_t* skip, bool* corrupt = nullptr) {
        uint64_t t = tail.load(std::memory_order_relaxed);
        uint64_t h = head.load(std::memory_order_acquire);
        if (h == t) return nullptr;
        uint32_t pos = static_cast<uint32_t>(t % cap);
        uint32_t until_end = cap - pos;
        uint32_t l;
        if (until_end < 4) {  // exhausted tail (no room for a marker)
            *len = 0;
            *skip = until_end;
            return data;  // non-null: caller consumes the skip and re-peeks
        }
        memcpy(&l, data + pos, 4);
        if (l == kWrapMarker) {
            *len = 0;
            *skip = until_end;
            return data;
        }
        if (l < sizeof(RecHdr) || l > until_end || l > h - t || (l & 7)) {
            if (corrupt) *corrupt = true;
            return nullptr;
        }
        *len = l;
        *skip = l;
        return data + pos;
    }
    void consume(uint64_t skip) {
        tail.store(tail.load(std::memory_order_relaxed) + skip, std::memory_order_release);
    }
};

// Segment layout: [Ctrl][req Ring][resp Ring].
struct Ctrl {
    uint64_t magic;
    uint32_t version;
    uint32_t req_off;
    uint32_t req_cap;
    uint32_t resp_off;
    uint32_t resp_cap;
    uint32_t _pad;
};

struct Segment {
    void* base = nullptr;
    size_t len = 0;
    Ring* req = nullptr;
    Ring* resp = nullptr;

    bool attach() {  // validate + wire pointers (base/len set by map)
        if (len < sizeof(Ctrl)) return false;
        auto* c = static_cast<Ctrl*>(base);
        if (c->magic != kMagic || c->version != kVersion) return false;
        if (c->req_cap == 0 || c->resp_cap == 0) return false;
        if ((c->req_cap | c->resp_cap) & 7) return false;
        // Ring::head/tail are std::atomic<uint64_t>: a hostile segment with
        // misaligned ring offsets would make those accesses UB / torn.
        if ((c->req_off | c->resp_off) & 7) return false;
        uint64_t req_end = uint64_t(c->req_off) + Ring::footprint(c->req_cap);
        uint64_t resp_end = uint64_t(c->resp_off) + Ring::footprint(c->resp_cap);
        if (c->req_off < sizeof(Ctrl) || req_end > len) return false;
        if (c->resp_off < req_end || resp_end > len) return false;
        req = reinterpret_cast<Ring*>(static_cast<uint8_t*>(base) + c->req_off);
        resp = reinterpret_cast<Ring*>(static_cast<uint8_t*>(base) + c->resp_off);
        return true;
    }

    void unmap() {
        if (base) munmap(base, len);
        base = nullptr;
        req = resp = nullptr;
    }
};

inline size_t segment_len(uint32_t req_cap, uint32_t resp_cap) {
    return sizeof(Ctrl) + Ring::footprint(req_cap) + Ring::footprint(resp_cap);
}

// Client side: create + initialize a segment under `name` (shm_open).
inline bool create_segment(const std::string& name, uint32_t req_cap, uint32_t resp_cap,
                           Segment* out) {
    int fd = shm_open(name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
    if (fd < 0) return false;
    size_t len = segment_len(req_cap, resp_cap);
    if (ftruncate(fd, static_cast<off_t>(len)) != 0) {
        ::close(fd);
        shm_unlink(name.c_str());
        return false;
    }
    void* p = mmap(nullptr, len, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    ::close(fd);
    if (p == MAP_FAILED) {
        shm_unlink(name.c_str());
        return false;
    }
    memset(p, 0, sizeof(Ctrl) + sizeof(Ring));
    auto* c = static_cast<Ctrl*>(p);
    c->req_off = sizeof(Ctrl);
    c->req_cap = req_cap;
    c->resp_off = static_cast<uint32_t>(sizeof(Ctrl) + Ring::footprint(req_cap));
    c->resp_cap = resp_cap;
    auto* req = reinterpret_cast<Ring*>(static_cast<uint8_t*>(p) + c->req_off);
    auto* resp = reinterpret_cast<Ring*>(static_cast<uint8_t*>(p) + c->resp_off);
    new (&req->head) std::atomic<uint64_t>(0);
    new (&req->tail) std::atomic<uint64_t>(0);
    req->cap = req_cap;
    new (&resp->head) std::atomic<uint64_t>(0);
    new (&resp->tail) std::atomic<uint64_t>(0);
    resp->cap = resp_cap;
    c->version = kVersion;
    std::atomic_thread_fence(std::memory_order_release);
    c->magic = kMagic;  // last: the server validates magic after mapping
    out->base = p;
    out->len = len;
    return out->attach();
}

// Server side: map an existing segment by name.
inline bool open_segment(const std::string& name, Segment* out) {
    int fd = shm_open(name.c_str(), O_RDWR, 0);
    if (fd < 0) return false;
    off_t sz = lseek(fd, 0, SEEK_END);
    if (sz < static_cast<off_t>(sizeof(Ctrl)) || sz > (64 << 20)) {
        ::close(fd);
        return false;
    }
    void* p = mmap(nullptr, static_cast<size_t>(sz), PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    ::close(fd);
    if (p == MAP_FAILED) return false;
    out->base = p;
    out->len = static_cast<size_t>(sz);
    if (!out->attach()) {
        out->unmap();
        return false;
    }
    return true;
}

}  // namespace shmring
}  // namespace ifs
