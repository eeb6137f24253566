// brpc_amd: Streaming RPC — ordered, flow-controlled byte streams
// established through a normal RPC and multiplexed on the same socket.
// Parity: reference brpc/stream.h (StreamCreate/StreamAccept/StreamWrite/
// StreamClose/StreamWait, StreamOptions{max_buf_size, handler}) +
// policy/streaming_rpc_protocol.cpp (DATA/CLOSE/FEEDBACK frames with
// consumed-size feedback windows).
//
// MI355X note: frames may carry HBM-resident IOBuf payloads; with 1 MiB
// frames between two GPUs the data path is RCCL p2p over xGMI in the bench
// harness, with this stream layer providing ordering + credit.
#pragma once

#include <functional>

#include "base/iobuf.h"

namespace bam {

class Controller;

typedef uint64_t StreamId;  // versioned; 0 = invalid

class CommGroup;

struct StreamOptions {
  size_t max_buf_size = 2u << 20;  // send window (bytes in flight)
  // Called in the stream's consumer fiber for each arrived message.
  std::function<void(StreamId, IOBuf* msg)> on_received;
  std::function<void(StreamId)> on_closed;
  // xGMI data plane (BASELINE config 3): when both ends set the same
  // (gpu_group, gpu_peer) pair, DATA messages whose payload is one
  // HBM-resident span travel as RCCL p2p over xGMI; the TCP frame carries
  // only a size descriptor (ordering + credit). The receiving side lands
  // the payload in a fresh HBM block (no host staging). Host-resident
  // messages on the same stream still go over TCP. gpu_peer is the PEER's
  // group rank. Not owned.
  CommGroup* gpu_group = nullptr;
  int gpu_peer = -1;
};

// Client side: create before CallMethod; the stream id rides in the
// request meta. After the RPC succeeds the stream is connected.
int StreamCreate(StreamId* sid, Controller* cntl, const StreamOptions& opt);

// Server side: accept inside the service handler (request meta must carry
// a stream id). The response meta carries our id back.
int StreamAccept(StreamId* sid, Controller* cntl, const StreamOptions& opt);

// Writes one message. Blocks the calling fiber while the send window is
// full. Returns 0, EINVAL (bad/closed stream), EPIPE (connection broken).
int StreamWrite(StreamId sid, IOBuf* data);

// Sends CLOSE and tears down locally. Idempotent.
int StreamClose(StreamId sid);

// Blocks until the stream is closed (by either side).
int StreamWait(StreamId sid);

bool StreamExists(StreamId sid);

// internal: protocol + plumbing hooks
namespace stream_internal {
void OnStreamFrame(uint64_t dst_sid, int type, uint64_t aux, IOBuf* payload,
                   uint64_t socket_id);
// Connects a client-created stream to the peer once the RPC response
// arrives (remote_sid from response meta).
int ConnectLocalStream(StreamId local, uint64_t remote_sid, uint64_t socket_id);
// Registers protocol parser (idempotent).
void RegisterStreamProtocol();
}  // namespace stream_internal

}  // namespace bam
