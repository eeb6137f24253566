#include "rpc/policy/std_protocol.h"

#include <mutex>

#include "base/logging.h"
#include "base/time.h"
#include "rpc/authenticator.h"
#include "rpc/channel.h"
#include "rpc/controller.h"
#include "rpc/concurrency_limiter.h"
#include "rpc/server.h"
#include "rpc/compress.h"
#include "rpc/policy/http_protocol.h"
#include "rpc/rpc_dump.h"
#include "rpc/rpcz.h"
#include "rpc/stream.h"
#include "rpc/wire.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

namespace policy {

static const char kMagic[4] = {'P', 'R', 'P', 'C'};
static const size_t kHeaderLen = 12;

// ---------------- meta codec ----------------

void SerializeRpcMeta(const RpcMeta& meta, std::string* out) {
  if (meta.has_request) {
    std::string sub;
    wire::put_str_field(&sub, 1, meta.service_name);
    wire::put_str_field(&sub, 2, meta.method_name);
    if (meta.log_id != 0) wire::put_int_field(&sub, 3, (int64_t)meta.log_id);
    if (meta.trace_id != 0) wire::put_int_field(&sub, 4, (int64_t)meta.trace_id);
    if (meta.span_id != 0) wire::put_int_field(&sub, 5, (int64_t)meta.span_id);
    if (meta.parent_span_id != 0)
      wire::put_int_field(&sub, 6, (int64_t)meta.parent_span_id);
    wire::put_msg_field(out, 1, sub);
  }
  if (meta.has_response) {
    std::string sub;
    if (meta.error_code != 0) wire::put_int_field(&sub, 1, meta.error_code);
    if (!meta.error_text.empty()) wire::put_str_field(&sub, 2, meta.error_text);
    wire::put_msg_field(out, 2, sub);
  }
  if (meta.compress_type != 0) wire::put_int_field(out, 3, meta.compress_type);
  wire::put_int_field(out, 4, meta.correlation_id);
  if (meta.attachment_size != 0) wire::put_int_field(out, 5, meta.attachment_size);
  if (!meta.auth_data.empty()) wire::put_str_field(out, 7, meta.auth_data);
  if (meta.stream_id != 0) {
    std::string sub;
    wire::put_int_field(&sub, 1, (int64_t)meta.stream_id);
    wire::put_msg_field(out, 8, sub);
  }
}

bool ParseRpcMeta(const char* data, size_t n, RpcMeta* out) {
  wire::Reader r(data, n);
  int wtype;
  for (int field; (field = r.read_tag(&wtype)) != 0;) {
    switch (field) {
      case 1: {  // RpcRequestMeta
        std::string sub = r.read_string();
        if (!r.ok()) return false;
        out->has_request = true;
        wire::Reader rr(sub.data(), sub.size());
        int wt2;
        for (int f2; (f2 = rr.read_tag(&wt2)) != 0;) {
          if (f2 == 1) out->service_name = rr.read_string();
          else if (f2 == 2) out->method_name = rr.read_string();
          else if (f2 == 3) out->log_id = rr.varint();
          else if (f2 == 4) out->trace_id = rr.varint();
          else if (f2 == 5) out->span_id = rr.varint();
          else if (f2 == 6) out->parent_span_id = rr.varint();
          else rr.skip(wt2);
          if (!rr.ok()) return false;
        }
        break;
      }
      case 2: {  // RpcResponseMeta
        std::string sub = r.read_string();
        if (!r.ok()) return false;
        out->has_response = true;
        wire::Reader rr(sub.data(), sub.size());
        int wt2;
        for (int f2; (f2 = rr.read_tag(&wt2)) != 0;) {
          if (f2 == 1) out->error_code = (int)rr.varint();
          else if (f2 == 2) out->error_text = rr.read_string();
          else rr.skip(wt2);
          if (!rr.ok()) return false;
        }
        break;
      }
      case 3:
        out->compress_type = (int)r.varint();
        break;
      case 4:
        out->correlation_id = (int64_t)r.varint();
        break;
      case 5:
        out->attachment_size = (int32_t)r.varint();
        break;
      case 7:
        out->auth_data = r.read_string();
        break;
      case 8: {
        std::string sub = r.read_string();
        if (!r.ok()) return false;
        wire::Reader rr(sub.data(), sub.size());
        int wt2;
        for (int f2; (f2 = rr.read_tag(&wt2)) != 0;) {
          if (f2 == 1) out->stream_id = rr.varint();
          else rr.skip(wt2);
          if (!rr.ok()) return false;
        }
        break;
      }
      default:
        r.skip(wtype);
    }
    if (!r.ok()) return false;
  }
  return r.ok();
}

// ---------------- framing ----------------

struct StdMessage : public InputMessageBase {
  RpcMeta meta;
  IOBuf payload;  // user data (+ attachment at tail)
};

static ParseResult ParseStdMessage(IOBuf* source, Socket* /*sock*/, bool /*read_eof*/) {
  char aux[kHeaderLen];
  if (source->size() < kHeaderLen) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(aux, kHeaderLen);
  if (h == nullptr || memcmp(h, kMagic, 4) != 0)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  uint32_t body_size = wire::get_u32_be(h + 4);
  uint32_t meta_size = wire::get_u32_be(h + 8);
  if (meta_size > body_size || body_size > (256u << 20))
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  if (source->size() < kHeaderLen + body_size)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  source->pop_front(kHeaderLen);
  std::string meta_bytes;
  source->cutn(&meta_bytes, meta_size);
  StdMessage* msg = new StdMessage;
  if (!ParseRpcMeta(meta_bytes.data(), meta_bytes.size(), &msg->meta)) {
    delete msg;
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  }
  source->cutn(&msg->payload, body_size - meta_size);
  return ParseResult::make_ok(msg);
}

void PackStdRequest(IOBuf* out, Controller* cntl, SessionId correlation_id) {
  RpcMeta meta;
  meta.has_request = true;
  meta.service_name = cntl->call.service_name;
  meta.method_name = cntl->call.method_name;
  meta.log_id = cntl->log_id();
  meta.trace_id = cntl->trace_id();
  meta.span_id = cntl->span_id();
  meta.parent_span_id = cntl->parent_span_id();
  meta.compress_type = (int)cntl->request_compress_type();
  meta.correlation_id = (int64_t)correlation_id;
  meta.attachment_size = (int32_t)cntl->request_attachment().size();
  meta.stream_id = cntl->call.stream_id;
  meta.auth_data = cntl->call.auth_data;
  // Compress the serialized request (NOT the attachment — it is a raw
  // pass-through by contract, like the reference's baidu_std attachment).
  IOBuf body_buf;
  if (cntl->request_compress_type() != COMPRESS_TYPE_NONE) {
    RegisterBuiltinCompressHandlers();
    if (!ApplyCompress(cntl->request_compress_type(), cntl->call.request_buf, &body_buf)) {
      meta.compress_type = 0;
      body_buf = cntl->call.request_buf;
    }
  } else {
    body_buf = cntl->call.request_buf;
  }
  std::string meta_bytes;
  SerializeRpcMeta(meta, &meta_bytes);
  size_t body = meta_bytes.size() + body_buf.size() + cntl->request_attachment().size();
  char header[kHeaderLen];
  memcpy(header, kMagic, 4);
  wire::put_u32_be(header + 4, (uint32_t)body);
  wire::put_u32_be(header + 8, (uint32_t)meta_bytes.size());
  out->append(header, kHeaderLen);
  out->append(meta_bytes);
  out->append(std::move(body_buf));
  out->append(cntl->request_attachment());   // zero-copy ref share
}

static void PackStdResponse(IOBuf* out, int64_t correlation_id, int error_code,
                            const std::string& error_text, const IOBuf& payload,
                            const IOBuf& attachment, uint64_t stream_id,
                            int compress_type = 0) {
  RpcMeta meta;
  meta.has_response = true;
  meta.error_code = error_code;
  meta.error_text = error_text;
  meta.correlation_id = correlation_id;
  meta.attachment_size = (int32_t)attachment.size();
  meta.stream_id = stream_id;
  meta.compress_type = compress_type;
  std::string meta_bytes;
  SerializeRpcMeta(meta, &meta_bytes);
  size_t body = meta_bytes.size() + payload.size() + attachment.size();
  char header[kHeaderLen];
  memcpy(header, kMagic, 4);
  wire::put_u32_be(header + 4, (uint32_t)body);
  wire::put_u32_be(header + 8, (uint32_t)meta_bytes.size());
  out->append(header, kHeaderLen);
  out->append(meta_bytes);
  out->append(payload);
  out->append(attachment);
}

// ---------------- server side ----------------

static void SendStdResponse(SocketId sid, int64_t cid, Controller* cntl, IOBuf* resp,
                            var::LatencyRecorder* status, int64_t start_us) {
  SocketUniquePtr sock;
  if (Socket::Address(sid, &sock) == 0) {
    IOBuf body = *resp;
    int resp_compress = 0;
    if (cntl->response_compress_ != COMPRESS_TYPE_NONE && !cntl->Failed()) {
      RegisterBuiltinCompressHandlers();
      IOBuf compressed;
      if (ApplyCompress(cntl->response_compress_, body, &compressed)) {
        body.swap(compressed);
        resp_compress = (int)cntl->response_compress_;
      }
    }
    IOBuf packet;
    PackStdResponse(&packet, cid, cntl->ErrorCode(), cntl->ErrorText(), body,
                    cntl->response_attachment(), cntl->response_stream_id_, resp_compress);
    sock->Write(&packet);
  }
  if (cntl->server_ != nullptr) {
    // Only requests that reached the handler were counted in (early
    // failures — unknown method, ELIMIT, auth — never incremented).
    if (cntl->concurrency_counted_) {
      cntl->server_->concurrency.fetch_sub(1, std::memory_order_relaxed);
    }
    cntl->server_->nprocessed.fetch_add(1, std::memory_order_relaxed);
  }
  if (status != nullptr) *status << (monotonic_time_us() - start_us);
  if (cntl->server_ != nullptr && cntl->server_->limiter() != nullptr &&
      cntl->concurrency_counted_) {
    cntl->server_->limiter()->OnResponse(cntl->ErrorCode(), monotonic_time_us() - start_us);
  }
  if (cntl->server_ != nullptr && cntl->method_gate_entered_) {
    cntl->server_->EndMethod(cntl->call.service_name, cntl->call.method_name);
  }
  if (rpcz::enabled()) {
    rpcz::Span span;
    span.start_us = start_us;
    span.end_us = monotonic_time_us();
    span.full_method = cntl->call.service_name + "." + cntl->call.method_name;
    span.remote = cntl->remote_side();
    span.error_code = cntl->ErrorCode();
    span.log_id = cntl->log_id();
    span.trace_id = cntl->trace_id();
    span.span_id = cntl->span_id();
    span.parent_span_id = cntl->parent_span_id();
    span.server_side = true;
    span.response_size = resp->size();
    rpcz::RecordSpan(span);
  }
  delete resp;
  delete cntl;
}

static void ProcessStdRequest(InputMessageBase* msg_base) {
  StdMessage* msg = (StdMessage*)msg_base;
  SocketUniquePtr sock;
  if (Socket::Address(msg->socket_id, &sock) != 0) {
    delete msg;
    return;
  }
  Server* server = (Server*)sock->user();
  Controller* cntl = new Controller;
  cntl->server_ = server;
  cntl->server_socket_ = sock->id();
  cntl->server_cid_ = msg->meta.correlation_id;
  cntl->log_id_ = msg->meta.log_id;
  cntl->trace_id_ = msg->meta.trace_id;
  cntl->span_id_ = msg->meta.span_id;
  cntl->parent_span_id_ = msg->meta.parent_span_id;
  cntl->call.service_name = msg->meta.service_name;
  cntl->call.method_name = msg->meta.method_name;
  cntl->remote_stream_id_ = msg->meta.stream_id;
  cntl->remote_side_ = sock->remote_side();
  cntl->local_side_ = sock->local_side();
  // Per-connection authentication (verified once, cached on the socket).
  if (server != nullptr && server->options().auth != nullptr) {
    AuthContext* ctx = sock->auth_context();
    if (ctx == nullptr) {
      AuthContext* fresh = new AuthContext;
      if (server->options().auth->VerifyCredential(msg->meta.auth_data, sock->remote_side(),
                                                   fresh) != 0) {
        delete fresh;
        IOBuf* resp = new IOBuf;
        cntl->SetFailed(ERPCAUTH, "authentication failed: " +
                                      server->options().auth->GetUnauthorizedErrorText());
        SocketId sid0 = sock->id();
        int64_t cid0 = msg->meta.correlation_id;
        delete msg;
        SendStdResponse(sid0, cid0, cntl, resp, nullptr, monotonic_time_us());
        return;
      }
      if (!sock->set_auth_context(fresh)) delete fresh;  // lost install race
      ctx = sock->auth_context();
    }
    cntl->auth_context_ = ctx;
  }
  IOBuf* resp = new IOBuf;
  SocketId sid = sock->id();
  int64_t cid = msg->meta.correlation_id;
  int64_t start_us = monotonic_time_us();
  var::LatencyRecorder* status =
      server != nullptr ? server->method_status(msg->meta.service_name, msg->meta.method_name)
                        : nullptr;
  Closure* done = NewCallback(
      [sid, cid, cntl, resp, status, start_us] { SendStdResponse(sid, cid, cntl, resp, status, start_us); });

  // attachment split
  IOBuf req_data;
  size_t att = (size_t)msg->meta.attachment_size;
  size_t data_len = msg->payload.size() >= att ? msg->payload.size() - att : 0;
  msg->payload.cutn(&req_data, data_len);
  cntl->request_attachment().swap(msg->payload);
  if (msg->meta.compress_type != 0) {
    RegisterBuiltinCompressHandlers();
    IOBuf plain;
    if (!ApplyDecompress((CompressType)msg->meta.compress_type, req_data, &plain)) {
      cntl->SetFailed(EREQUEST, "request decompression failed");
      delete msg;
      done->Run();
      return;
    }
    req_data.swap(plain);
  }

  Service* svc = nullptr;
  const MethodFn* fn =
      server != nullptr
          ? server->FindMethod(msg->meta.service_name, msg->meta.method_name, &svc)
          : nullptr;
  if (fn == nullptr && server != nullptr && server->options().master_handler) {
    // Generic/proxy pass-through (≙ reference BaiduMasterService).
    fn = &server->options().master_handler;
  }
  if (fn == nullptr) {
    cntl->SetFailed(msg->meta.service_name.empty() || svc == nullptr ? ENOSERVICE : ENOMETHOD,
                    "unknown service/method " + msg->meta.service_name + "." +
                        msg->meta.method_name);
    delete msg;
    done->Run();
    return;
  }
  if (server->max_concurrency() > 0 &&
      server->concurrency.load(std::memory_order_relaxed) >= server->max_concurrency()) {
    cntl->SetFailed(ELIMIT, "reached server max_concurrency");
    delete msg;
    done->Run();
    return;
  }
  if (server->limiter() != nullptr &&
      !server->limiter()->OnRequest(server->concurrency.load(std::memory_order_relaxed) + 1)) {
    cntl->SetFailed(ELIMIT, "rejected by adaptive concurrency limiter");
    delete msg;
    done->Run();
    return;
  }
  if (!server->BeginMethod(msg->meta.service_name, msg->meta.method_name)) {
    cntl->SetFailed(ELIMIT, "reached method_max_concurrency of " +
                                msg->meta.service_name + "." + msg->meta.method_name);
    delete msg;
    done->Run();
    return;
  }
  cntl->method_gate_entered_ = true;
  if (server->options().interceptor) {
    int ec = 0;
    std::string etext;
    if (!server->options().interceptor(cntl, &ec, &etext)) {
      cntl->SetFailed(ec != 0 ? ec : EREQUEST, etext.empty() ? "rejected by interceptor" : etext);
      delete msg;
      done->Run();
      return;
    }
  }
  server->concurrency.fetch_add(1, std::memory_order_relaxed);
  cntl->concurrency_counted_ = true;
  rpc_dump::SampleRequest(msg->meta.service_name, msg->meta.method_name, req_data);
  // Export the inbound trace while the handler runs: nested client calls
  // made inside it chain parent_span_id from this server span. (Handlers
  // that stash `done` and return before issuing sub-calls fall outside
  // the ambient window — pass ids explicitly via set_trace_id there.)
  const bool ambient_trace = cntl->trace_id_ != 0;
  if (ambient_trace) rpcz::set_current_trace(cntl->trace_id_, cntl->span_id_);
  (*fn)(cntl, req_data, resp, done);
  if (ambient_trace) rpcz::clear_current_trace();
  delete msg;
}

// ---------------- client side ----------------

static void ProcessStdResponse(InputMessageBase* msg_base) {
  StdMessage* msg = (StdMessage*)msg_base;
  SessionId cid = (SessionId)msg->meta.correlation_id;
  void* data = nullptr;
  if (session_lock(cid, &data) != 0) {
    delete msg;  // late/duplicate response
    return;
  }
  Controller* cntl = (Controller*)data;
  // Responses from ANY live attempt complete the call (backup requests /
  // retries race; first response wins — parity with the reference's
  // versioned-correlation semantics).
  {
    SocketUniquePtr sock;
    if (Socket::Address(msg->socket_id, &sock) == 0) sock->remove_pending_session(cid);
  }
  if (msg->meta.error_code != 0) {
    cntl->SetFailed(msg->meta.error_code, msg->meta.error_text);
  } else {
    if (msg->meta.stream_id != 0 && cntl->call.stream_id != 0) {
      stream_internal::ConnectLocalStream(cntl->call.stream_id, msg->meta.stream_id,
                                          msg->socket_id);
    }
    size_t att = (size_t)msg->meta.attachment_size;
    size_t data_len = msg->payload.size() >= att ? msg->payload.size() - att : 0;
    if (cntl->call.response != nullptr) {
      cntl->call.response->clear();
      msg->payload.cutn(cntl->call.response, data_len);
      if (msg->meta.compress_type != 0) {
        RegisterBuiltinCompressHandlers();
        IOBuf plain;
        if (ApplyDecompress((CompressType)msg->meta.compress_type, *cntl->call.response,
                            &plain)) {
          cntl->call.response->swap(plain);
        } else {
          cntl->SetFailed(ERESPONSE, "response decompression failed");
        }
      }
    } else {
      msg->payload.pop_front(data_len);
    }
    cntl->response_attachment().clear();
    cntl->response_attachment().swap(msg->payload);
  }
  delete msg;
  EndRPC(cntl, cid);
}

void RegisterStdProtocol() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    Protocol p;
    p.parse = ParseStdMessage;
    p.process_request = ProcessStdRequest;
    p.process_response = ProcessStdResponse;
    p.support_server = true;
    p.support_client = true;
    p.name = "std";
    RegisterProtocol(p);
    stream_internal::RegisterStreamProtocol();
    RegisterHttpProtocol();
  });
}

}  // namespace policy
}  // namespace bam
