// brpc_amd: RTMP client (publisher / player).
// Parity: reference brpc/rtmp.h RtmpClientStream-family (clean-room
// subset): plain handshake, connect(app), createStream, publish or play,
// then media frames (audio/video/data) in either direction over the chunk
// stream (rpc/rtmp_chunk.h). Blocking-socket client like MysqlClient.
#pragma once

#include <stdint.h>

#include <string>

#include "rpc/rtmp_chunk.h"

namespace bam {

class RtmpClient {
 public:
  ~RtmpClient();

  // Handshake + connect(app). 0 on success.
  int Connect(const std::string& host, int port, const std::string& app,
              int timeout_ms = 3000);
  // createStream + publish(name). 0 on success.
  int Publish(const std::string& stream_name);
  // createStream + play(name). 0 on success.
  int Play(const std::string& stream_name);

  // Publisher: send one media message (type 8 audio / 9 video / 18 data).
  int PushFrame(uint8_t type, uint32_t timestamp, const std::string& payload);

  // Player: wait for the next audio/video/data message.
  // 0 on success, -1 on timeout/transport error.
  int PollFrame(rtmp::Message* out, int timeout_ms = 3000);

  void Close();
  bool connected() const { return fd_ >= 0; }

 private:
  int send_bytes(const std::string& bytes);
  int read_more(int timeout_ms);
  // Reads messages until `command` result arrives (returns its payload
  // values decoded ok) or timeout.
  int wait_command_reply(const std::string& expect_cmd, int timeout_ms);
  int send_command(const std::string& payload_amf, uint32_t msid = 0);

  int fd_ = -1;
  int timeout_ms_ = 3000;
  rtmp::ChunkReader reader_;
  uint32_t out_chunk_ = 128;  // until the server's SetChunkSize arrives (we keep 128 out)
  double next_txn_ = 1;
};

}  // namespace bam
