// brpc_amd: SocketMap — client-side connection dedup per EndPoint.
// Parity: reference brpc/socket_map.h (single connection type).
#pragma once

#include "base/endpoint.h"
#include "rpc/socket.h"

namespace bam {

// Returns a referenced, healthy client socket to `ep`, creating/connecting
// if needed. 0 on success. `shard` > 0 selects a distinct pooled
// connection (parity: reference pooled connection_type — multiple
// connections to one server spread parse/write parallelism).
int GetClientSocket(const EndPoint& ep, SocketUniquePtr* out, int shard = 0,
                    bool use_breaker = true,
                    bool ssl = false, int protocol_index = -1,
                    const char* socket_mode = nullptr /* "rdma_mock" */,
                    bool short_conn = false /* fresh socket, never pooled */);

// Drops the cached socket for ep (e.g. after failure).
void RemoveClientSocket(const EndPoint& ep, SocketId expected);

// ---- circuit breaker + health check ----
// Parity: reference brpc/circuit_breaker.{h,cpp} + details/health_check.cpp:
// consecutive network failures isolate the endpoint (GetClientSocket fails
// fast with EHOSTDOWN); a background fiber re-probes every
// -health_check_interval_ms and revives on a successful connect.
void ReportClientCallResult(const EndPoint& ep, bool network_error);
bool IsEndpointIsolated(const EndPoint& ep);

}  // namespace bam
