// Small shared helpers (socket exact-IO, refcount base).
// Same roles as /root/reference/src/utils.h:19-56 — fresh implementation.
#pragma once

#include <atomic>
#include <cerrno>
#include <cstddef>
#include <cstdint>
#include <sys/socket.h>
#include <unistd.h>

namespace ifs {

// Blocking exact send/recv; return false on error/EOF.
inline bool send_exact(int fd, const void* buf, size_t n) {
    const uint8_t* p = static_cast<const uint8_t*>(buf);
    while (n) {
        ssize_t r = ::send(fd, p, n, MSG_NOSIGNAL);
        if (r < 0) {
            if (errno == EINTR) continue;
            return false;
        }
        if (r == 0) return false;
        p += r;
        n -= static_cast<size_t>(r);
    }
    return true;
}

inline bool recv_exact(int fd, void* buf, size_t n) {
    uint8_t* p = static_cast<uint8_t*>(buf);
    while (n) {
        ssize_t r = ::recv(fd, p, n, 0);
        if (r < 0) {
            if (errno == EINTR) continue;
            return false;
        }
        if (r == 0) return false;
        p += r;
        n -= static_cast<size_t>(r);
    }
    return true;
}

// Intrusive refcount base for pool block handles (kv index entries keep
// blocks alive while reads are in flight; cf. reference utils.h:35-56).
class RefCounted {
   public:
    RefCounted() = default;
    RefCounted(const RefCounted&) = delete;
    RefCounted& operator=(const RefCounted&) = delete;
    virtual ~RefCounted() = default;

    void ref() const { count_.fetch_add(1, std::memory_order_relaxed); }
    void unref() const {
        if (count_.fetch_sub(1, std::memory_order_acq_rel) == 1) delete this;
    }
    int ref_count() const { return count_.load(std::memory_order_relaxed); }

   private:
    mutable std::atomic<int> count_{1};
};

template <typename T>
class Ref {
   public:
    Ref() : p_(nullptr) {}
    explicit Ref(T* p) : p_(p) {}  // adopts (no extra ref)
    Ref(const Ref& o) : p_(o.p_) {
        if (p_) p_->ref();
    }
    Ref(Ref&& o) noexcept : p_(o.p_) { o.p_ = nullptr; }
    Ref& operator=(const Ref& o) {
        if (this != &o) {
            if (o.p_) o.p_->ref();
            if (p_) p_->unref();
            p_ = o.p_;
        }
        return *this;
    }
    Ref& operator=(Ref&& o) noexcept {
        if (this != &o) {
            if (p_) p_->unref();
            p_ = o.p_;
            o.p_ = nullptr;
        }
        return *this;
    }
    ~Ref() {
        if (p_) p_->unref();
    }
    T* get() const { return p_; }
    T* operator->() const { return p_; }
    T& operator*() const { return *p_; }
    explicit operator bool() const { return p_ != nullptr; }

   private:
    T* p_;
};

}  // namespace ifs
