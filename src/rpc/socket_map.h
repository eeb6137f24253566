// brpc_amd: SocketMap — client-side connection dedup per EndPoint.
// Parity: reference brpc/socket_map.h (single connection type).
#pragma once

#include "base/endpoint.h"
#include "rpc/socket.h"

namespace bam {

// Returns a referenced, healthy client socket to `ep`, creating/connecting
// if needed. 0 on success.
int GetClientSocket(const EndPoint& ep, SocketUniquePtr* out);

// Drops the cached socket for ep (e.g. after failure).
void RemoveClientSocket(const EndPoint& ep, SocketId expected);

}  // namespace bam
