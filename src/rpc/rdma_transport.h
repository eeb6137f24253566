// brpc_amd: RDMA transport behind the Transport seam.
// Parity: reference brpc/rdma/rdma_endpoint.cpp:714-998 (credit-window
// send/recv over posted fixed-size registered blocks, imm-carried credit
// returns) + rdma/block_pool.cpp:56 (registered {8K,64K,2M} pools — here
// the pool classes mirror the HBM pool in hip/gpu_runtime.hip so
// GPUDirect lands payloads straight into HBM-backed registered memory).
//
// The verbs dependency is a narrow PROVIDER seam (RdmaProvider):
//   * "verbs"  — compiled only when <infiniband/verbs.h> exists on the
//     build host (this image has no libibverbs; the endpoint machinery is
//     still fully built and tested),
//   * "mock"   — in-process pairing keyed by the TCP connection's port
//     pair: two loopback sockets upgrade to paired endpoints and move
//     bytes through posted blocks with REAL credit accounting. This is
//     the test vehicle for the whole endpoint state machine, and the
//     shape a future xGMI peer-HBM provider plugs into.
//
// Wire protocol stays byte-stream: each PostSend moves one block-sized
// chunk; QP ordering preserves the stream, so every TCP protocol
// (baidu_std, http, ...) runs unchanged above it.
#pragma once

#include <stdint.h>

#include <string>

#include "rpc/transport.h"

namespace bam {

class Socket;

namespace rdma {

// Completion sink implemented by the endpoint.
class CompletionSink {
 public:
  virtual ~CompletionSink() {}
  // A peer message landed in `buf` (one of our posted recv blocks).
  // imm carries the peer's credit return count.
  virtual void OnRecv(char* buf, uint32_t len, uint32_t imm) = 0;
  virtual void OnSendDone(const char* buf) = 0;
  virtual void OnChannelBroken() = 0;
};

// The narrow verbs seam. One provider per process.
class RdmaProvider {
 public:
  virtual ~RdmaProvider() {}
  virtual const char* name() const = 0;
  // Connection-level handle; `key` identifies the peer (mock: the
  // normalized loopback port pair; verbs: from the TCP handshake).
  virtual void* CreateChannel(uint64_t key, CompletionSink* sink) = 0;
  virtual void DestroyChannel(void* ch) = 0;
  // Registers a DMA-able region (no-op lkey for mock).
  virtual int RegisterMemory(void* addr, size_t len, uint32_t* lkey) = 0;
  // Posts one ordered message send with an immediate. 0 ok, EAGAIN-style
  // nonzero when the channel is not yet paired (caller retries).
  virtual int PostSend(void* ch, const char* data, uint32_t len, uint32_t imm) = 0;
  // Hands the channel one recv block (completions via sink->OnRecv).
  virtual int PostRecv(void* ch, char* buf, uint32_t cap) = 0;
};

// Process-wide providers.
RdmaProvider* mock_provider();
RdmaProvider* verbs_provider();  // nullptr when verbs is unavailable

// Creates the RDMA transport for `socket` (called after the TCP
// connection exists; the port pair keys mock pairing). Returns nullptr
// +err when the provider is unavailable.
Transport* CreateRdmaTransport(Socket* socket, RdmaProvider* provider,
                               uint32_t window_blocks, uint32_t block_bytes,
                               std::string* err);

// Diagnostics for tests: blocks currently owned by live endpoints.
int64_t live_recv_blocks();

}  // namespace rdma
}  // namespace bam
