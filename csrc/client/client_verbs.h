// Client-side verbs data plane (see verbs_fabric.h for availability
// caveats): QP bring-up mirroring the server, allocate over IBV_WR_SEND,
// one-sided RDMA_WRITE puts with a commit SEND, reads completed by the
// server's WRITE_WITH_IMM into a pre-posted zero-length recv — the
// reference's client flows (libinfinistore.cpp:285-430, 748-1099).
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <utility>
#include <vector>

#include "../core/protocol.h"

namespace ifs {

struct ClientConfigC;

class VerbsClient {
   public:
    // Performs the OP_RDMA_EXCHANGE handshake over the connected TCP fd.
    // Returns nullptr when verbs is unavailable locally or the server
    // answered with the TCP fabric (caller falls back; `*attempted` tells
    // whether an exchange was already consumed on the socket).
    static std::unique_ptr<VerbsClient> establish(int fd, const ClientConfigC& cfg,
                                                  bool* attempted);
    ~VerbsClient();

    bool register_mr(void* addr, size_t len, bool device_mem);
    std::vector<std::pair<uint32_t, uint64_t>> allocate(const std::vector<std::string>& keys,
                                                        int block_size);
    // Blocking one-sided writes + commit; returns 0 on success.
    int write_blocks(const uint64_t* offsets, size_t n_offsets, int block_size,
                     const std::pair<uint32_t, uint64_t>* blocks, size_t n_blocks,
                     uintptr_t base_ptr);
    // Blocking read (server pushes; completes on WRITE_WITH_IMM).
    int read_blocks(const std::vector<std::pair<std::string, uint64_t>>& blocks,
                    int block_size, uintptr_t base_ptr);

   private:
    VerbsClient();
    struct Impl;
    std::unique_ptr<Impl> impl_;
};

}  // namespace ifs
