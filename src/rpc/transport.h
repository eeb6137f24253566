// brpc_amd: Transport — the byte-transport seam under Socket.
// Parity: reference brpc/transport.h:26-52 (CutFromIOBufList /
// WaitEpollOut / ProcessEvent) + transport_factory.cpp. The default TCP
// path stays inlined in Socket (exactly the reference's
// `_conn ? CutMessageIntoFileDescriptor : _transport->CutFromIOBuf`
// dispatch, socket.cpp:1752) so plugging a transport costs nothing when
// none is set; RDMA (rdma_transport.h) and future xGMI endpoints implement
// this interface.
#pragma once

#include <stdint.h>
#include <sys/types.h>

#include "base/iobuf.h"

namespace bam {

class Socket;

class Transport {
 public:
  virtual ~Transport() {}
  virtual const char* name() const = 0;

  // Writes bytes from the front of *data to the wire and pops what was
  // written. Returns bytes written; -1 with errno (EAGAIN = flow
  // controlled — caller waits via WaitWritable and retries).
  virtual ssize_t CutFromIOBuf(Socket* s, IOBuf* data) = 0;

  // Moves received bytes into *out (up to max). Returns bytes appended;
  // 0 with errno=EAGAIN when nothing is pending; 0 with errno=0 on EOF;
  // -1 on error.
  virtual ssize_t AppendToIOBuf(Socket* s, IOBuf* out, size_t max) = 0;

  // Blocks the calling fiber until the transport is writable again
  // (EPOLLOUT for TCP, returned credits for RDMA). 0 = writable,
  // -1 = failed/timeout.
  virtual int WaitWritable(Socket* s, int64_t abstime_us) = 0;
};

}  // namespace bam
