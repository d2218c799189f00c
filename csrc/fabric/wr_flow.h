// Work-request flow control for the RDMA data plane (transport-agnostic,
// unit-tested on CPU).
//
// Same flow-control roles as the reference (protocol.h:23-34 and the
// chain/overflow logic in infinistore.cpp:456-530 / libinfinistore.cpp:
// 898-987): writes are posted in chains of at most kMaxWrBatch WRs with one
// signaled completion per chain, at most kMaxOutstandingWrites WRs
// outstanding; chains that would exceed the cap are parked in an overflow
// queue and drained as completions arrive.
#pragma once

#include <algorithm>
#include <cstddef>
#include <cstdint>
#include <deque>
#include <functional>
#include <vector>

#include "../core/protocol.h"

namespace ifs {

struct WrDesc {
    uint64_t local_addr;
    uint64_t remote_addr;
    uint32_t len;
    uint32_t lkey;
    uint32_t rkey;
};

// A chain of <= kMaxWrBatch writes, posted as one linked ibv_post_send (the
// last WR is signaled; optionally carries an immediate for the peer).
struct WrChain {
    std::vector<WrDesc> wrs;
    bool with_imm = false;
    uint32_t imm_data = 0;
    uint64_t signal_cookie = 0;  // returned to on_chain_complete
};

// Splits descriptors into chains and meters them against the outstanding-WR
// budget. The transport supplies post_fn (actually posts a chain; returns
// false on transport error) and calls on_completion(n_wrs_of_chain) from its
// CQ handler; the meter then drains parked chains.
class WrFlow {
   public:
    using PostFn = std::function<bool(const WrChain&)>;

    explicit WrFlow(PostFn post, int max_batch = kMaxWrBatch,
                    int max_outstanding = kMaxOutstandingWrites)
        : post_(std::move(post)), max_batch_(max_batch), max_outstanding_(max_outstanding) {}

    // Enqueue a logical write set; the final chain of the set carries
    // (with_imm, imm, cookie) so the peer/completion sees the set boundary.
    bool submit(std::vector<WrDesc> wrs, bool with_imm, uint32_t imm, uint64_t cookie) {
        for (size_t off = 0; off < wrs.size(); off += static_cast<size_t>(max_batch_)) {
            size_t take = std::min(static_cast<size_t>(max_batch_), wrs.size() - off);
            WrChain ch;
            ch.wrs.assign(wrs.begin() + static_cast<long>(off),
                          wrs.begin() + static_cast<long>(off + take));
            bool last = off + take >= wrs.size();
            ch.with_imm = last && with_imm;
            ch.imm_data = last ? imm : 0;
            ch.signal_cookie = last ? cookie : 0;
            if (!push_chain(std::move(ch))) return false;
        }
        // A zero-WR set with an immediate still needs a bare IMM chain.
        if (wrs.empty() && with_imm) {
            WrChain ch;
            ch.with_imm = true;
            ch.imm_data = imm;
            ch.signal_cookie = cookie;
            return push_chain(std::move(ch));
        }
        return true;
    }

    // CQ handler reports one chain completion (n = WRs that chain held).
    bool on_chain_complete(size_t n_wrs) {
        outstanding_ -= static_cast<long>(n_wrs);
        // Drain parked chains while budget allows.
        while (!parked_.empty() &&
               outstanding_ + static_cast<long>(parked_.front().wrs.size()) <=
                   max_outstanding_) {
            WrChain ch = std::move(parked_.front());
            parked_.pop_front();
            outstanding_ += static_cast<long>(ch.wrs.size());
            if (!post_(ch)) return false;
        }
        return true;
    }

    long outstanding() const { return outstanding_; }
    size_t parked() const { return parked_.size(); }

   private:
    bool push_chain(WrChain&& ch) {
        if (outstanding_ + static_cast<long>(ch.wrs.size()) > max_outstanding_) {
            parked_.push_back(std::move(ch));
            return true;
        }
        outstanding_ += static_cast<long>(ch.wrs.size());
        return post_(ch);
    }

    PostFn post_;
    int max_batch_;
    int max_outstanding_;
    long outstanding_ = 0;
    std::deque<WrChain> parked_;
};

}  // namespace ifs
