// brpc_amd: Controller — per-RPC context for both client and server side.
// API parity: reference brpc/controller.h (set_timeout_ms / set_max_retry /
// request_attachment / response_attachment / Failed / ErrorCode/ErrorText /
// call_id / log_id / remote_side), payload-centric: request/response bodies
// are IOBufs (opaque bytes or serialized messages); attachments bypass
// serialization exactly like baidu_std's attachment field.
#pragma once

#include <stdint.h>

#include <map>
#include <memory>
#include <string>

#include "base/endpoint.h"
#include "base/iobuf.h"
#include "fiber/session.h"
#include "fiber/timer_thread.h"
#include "rpc/closure.h"
#include "rpc/rpc_errno.h"
#include "rpc/socket.h"

namespace bam {

enum CompressType {
  COMPRESS_TYPE_NONE = 0,
  COMPRESS_TYPE_SNAPPY = 1,
  COMPRESS_TYPE_GZIP = 2,
};

class Channel;
class Server;

// Minimal HTTP header view for http/h2 channels (≙ reference brpc
// HttpHeader used by Controller::http_request()/http_response(),
// controller.h:400-430): verb/content-type/status plus a lower-cased
// header map.
struct HttpHeaderExt {
  std::string method;        // client request: verb override (PUT, DELETE…)
  std::string content_type;  // both directions
  int status_code = 0;       // response side
  std::map<std::string, std::string> headers;  // lower-cased names

  void SetHeader(const std::string& k, const std::string& v);
  // nullptr when absent; name lookup is case-insensitive.
  const std::string* GetHeader(const std::string& k) const;
};

class Controller {
 public:
  Controller() { Reset(); }
  ~Controller();

  void Reset();

  // ---- client-side knobs ----
  void set_timeout_ms(int64_t ms) { timeout_ms_ = ms; }
  int64_t timeout_ms() const { return timeout_ms_; }
  void set_max_retry(int n) { max_retry_ = n; }
  int max_retry() const { return max_retry_; }
  void set_backup_request_ms(int64_t ms) { backup_request_ms_ = ms; }
  int64_t backup_request_ms() const { return backup_request_ms_; }
  void set_log_id(uint64_t id) { log_id_ = id; }
  uint64_t log_id() const { return log_id_; }

  // Consistent-hash routing key (≙ reference Controller::set_request_code,
  // controller.h: c_hash LBs route by this instead of a random pick).
  void set_request_code(uint64_t code) {
    request_code_ = code;
    has_request_code_ = true;
  }
  uint64_t request_code() const { return request_code_; }
  bool has_request_code() const { return has_request_code_; }

  // Trace ids (≙ reference brpc/span.h trace propagation): client calls
  // inherit the ambient trace (rpcz::current_trace) unless set explicitly;
  // servers read them off RpcRequestMeta and re-export while the handler
  // runs, so nested client calls chain parent_span_id automatically.
  void set_trace_id(uint64_t id) { trace_id_ = id; }
  uint64_t trace_id() const { return trace_id_; }
  uint64_t span_id() const { return span_id_; }
  uint64_t parent_span_id() const { return parent_span_id_; }
  void set_request_compress_type(CompressType t) { request_compress_ = t; }
  CompressType request_compress_type() const { return request_compress_; }

  IOBuf& request_attachment() { return request_attachment_; }
  IOBuf& response_attachment() { return response_attachment_; }

  // ---- status ----
  bool Failed() const { return error_code_ != 0; }
  int ErrorCode() const { return error_code_; }
  const std::string& ErrorText() const { return error_text_; }
  void SetFailed(int code, const std::string& reason) {
    error_code_ = code != 0 ? code : EINTERNAL;
    error_text_ = reason;
  }
  void SetFailed(const std::string& reason) { SetFailed(EINTERNAL, reason); }

  // ---- progressive response reading (parity: reference
  // response_read_progressively / ProgressiveReader, socket.h:662) ----
  // When set before CallMethod on an "http" channel, response BODY bytes
  // are delivered to the reader incrementally as they arrive from the
  // socket (chunk-by-chunk for chunked coding, read-by-read otherwise)
  // instead of buffering the whole body; done=true marks the last chunk.
  // The final response IOBuf stays empty.
  typedef std::function<void(const IOBuf& chunk, bool done)> ProgressiveReader;
  void response_read_progressively(ProgressiveReader r) {
    progressive_reader_ = std::move(r);
  }
  const ProgressiveReader& progressive_reader() const { return progressive_reader_; }

  SessionId call_id() const { return cid_; }
  // Client-side cancellation (parity: reference StartCancel/IsCanceled):
  // conducts ECANCELED through the call's session — done runs (or the
  // synchronous CallMethod returns) with ErrorCode()==ECANCELED unless a
  // response won the race.
  void StartCancel();
  bool IsCanceled() const { return error_code_ == ECANCELED_RPC; }
  int64_t latency_us() const { return end_us_ - start_us_; }
  int retried_count() const { return retry_count_; }

  EndPoint remote_side() const { return remote_side_; }
  EndPoint local_side() const { return local_side_; }

  bool is_server_side() const { return server_ != nullptr; }

  // Per-connection user state (reference Controller::session_local_data):
  // lazily created from ServerOptions::session_local_data_factory; same
  // pointer for every request on one connection. nullptr without a factory.
  void* session_local_data();
  // Per-worker user state (reference Controller::thread_local_data):
  // lazily created from ServerOptions::thread_local_data_factory in the
  // current execution context (fiber-local; TLS off-fiber).
  void* thread_local_data();

  // HTTP-specific request/response views (http/h2 protocol channels and
  // server handlers): lazily allocated; has_* avoids the allocation when
  // only probing.
  HttpHeaderExt& http_request();
  HttpHeaderExt& http_response();
  bool has_http_request() const { return http_request_ != nullptr; }
  bool has_http_response() const { return http_response_ != nullptr; }

  // ---- internals (channel / protocol / server plumbing) ----
  struct Call {
    SessionId cid = 0;              // whole-call session
    IOBuf request_buf;              // packed user payload (pre-header)
    IOBuf* response = nullptr;      // user's output buffer
    Closure* done = nullptr;        // nullptr = synchronous
    std::string service_name;
    std::string method_name;
    EndPoint server_ep;
    TimerId timeout_timer = 0;
    SocketId pending_socket = 0;    // socket holding the pending session
    class LoadBalancer* lb = nullptr;
    class SubChannelCtx* sub_ctx = nullptr;  // ParallelChannel bookkeeping
    uint64_t stream_id = 0;                  // client-created stream (StreamCreate)
    int protocol_index = -1;                 // wire protocol for this call
    std::atomic<uint64_t>* socket_cache = nullptr;  // channel's cached socket id
    int connection_shard = 0;                        // pooled connection index
    bool use_breaker = true;                         // ChannelOptions.enable_circuit_breaker
    bool short_conn = false;                         // connection_type == "short"
    uint64_t short_socket = 0;                       // to close at EndRPC
    const class Authenticator* auth = nullptr;       // from ChannelOptions
    std::string auth_data;                           // credential for this attempt
    bool ssl = false;                                // TLS client connection
    const char* socket_mode = nullptr;               // "" / "rdma_mock"
    std::function<bool(int, int)> retry_policy;      // from ChannelOptions
  };
  Call call;

  // server-side context
  Server* server_ = nullptr;
  SocketId server_socket_ = 0;
  uint64_t remote_stream_id_ = 0;    // stream id carried by the peer's meta
  uint64_t response_stream_id_ = 0;  // stream accepted by this server (StreamAccept)
  int64_t server_cid_ = 0;  // correlation id to echo back
  bool concurrency_counted_ = false;  // incremented server->concurrency (must decrement)
  bool method_gate_entered_ = false;  // passed BeginMethod (must EndMethod)
  // Connection's verified auth result (owned by the server Socket);
  // nullptr when the server has no authenticator. (rpc/authenticator.h)
  const class AuthContext* auth_context_ = nullptr;
  const class AuthContext* auth_context() const { return auth_context_; }
  CompressType response_compress_ = COMPRESS_TYPE_NONE;
  void set_response_compress_type(CompressType t) { response_compress_ = t; }

  int64_t start_us_ = 0;
  int64_t end_us_ = 0;
  int error_code_ = 0;
  std::string error_text_;
  int64_t timeout_ms_ = -1;
  int64_t backup_request_ms_ = -1;
  int max_retry_ = 3;
  int retry_count_ = 0;
  uint64_t log_id_ = 0;
  uint64_t request_code_ = 0;
  bool has_request_code_ = false;
  std::unique_ptr<HttpHeaderExt> http_request_;
  std::unique_ptr<HttpHeaderExt> http_response_;
  uint64_t trace_id_ = 0;
  uint64_t span_id_ = 0;
  uint64_t parent_span_id_ = 0;
  CompressType request_compress_ = COMPRESS_TYPE_NONE;
  IOBuf request_attachment_;
  IOBuf response_attachment_;
  ProgressiveReader progressive_reader_;
  EndPoint remote_side_;
  EndPoint local_side_;
  SessionId cid_ = 0;
};

// Waits until the RPC identified by `id` completes (client-side sync call).
inline int Join(SessionId id) { return session_join(id); }

}  // namespace bam
