// brpc_amd: same-host shared-memory RPC transport.
// Parity: reference UBRing shared-memory RPC (docs/en/ubring.md:9-13 —
// "microsecond-level latency, millions of RPC/s"). Clean-room design:
// one POSIX shm segment per connection holding two SPSC byte rings
// (client->server requests, server->client responses). Records are
// length-prefixed and sequenced; the consumer spins briefly then naps
// (fiber_usleep), so wakeup latency stays in the low microseconds without
// pinning a core when idle.
//
// This is the host-side analogue of a device peer-ring: the ring layout
// (power-of-2 byte ring, 64 B aligned records, acquire/release seq) is
// the same shape an xGMI HBM ring would use between GPU peers.
#pragma once

#include <stdint.h>

#include <functional>
#include <string>

#include "base/iobuf.h"

namespace bam {

class Server;

namespace shm {

// Serves requests arriving on shm connection `name` with `server`'s method
// map (service dispatch identical to the TCP path). Spawns a poller fiber;
// returns 0 on success. The segment is created here; clients connect after.
int ServeShm(const std::string& name, Server* server, uint32_t ring_bytes = 4u << 20);

// Stops serving `name` and unlinks the segment.
void StopShm(const std::string& name);

// Client connection to a served segment. Thread-safe for concurrent calls.
class ShmChannel {
 public:
  ~ShmChannel();
  int Init(const std::string& name);
  // Synchronous call (fiber-blocking). Returns 0 or an rpc_errno.
  int Call(const std::string& full_method, const IOBuf& request, IOBuf* response,
           int64_t timeout_us = 1000000, std::string* error_text = nullptr);

  struct Impl;  // public: the poll fiber entry needs it

 private:
  Impl* impl_ = nullptr;
};

}  // namespace shm
}  // namespace bam
