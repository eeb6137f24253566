// brpc_amd: HTTP/2 / gRPC CLIENT protocol (channel protocols "grpc"/"h2"),
// fully in-tree on rpc/policy/h2_session.* + hpack.* (round 1 dlopened
// libnghttp2; that dependency is gone).
// Parity: reference h2/gRPC client paths in policy/http2_rpc_protocol.cpp.
// One session per connection (socket protocol_ctx); requests map
// stream_id -> correlation session; responses complete on END_STREAM with
// grpc-status trailer handling. Interop-tested against the official grpc
// python SERVER (tests/test_grpc.py).
#include <map>
#include <mutex>
#include <vector>

#include "base/logging.h"
#include "fiber/session.h"
#include "rpc/channel.h"
#include "rpc/controller.h"
#include "rpc/policy/h2_session.h"
#include "rpc/server.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

namespace policy {

namespace {

struct H2ClientStream {
  uint64_t cid = 0;
  IOBuf body;
  int grpc_status = -1;
  std::string grpc_message;
  int http_status = 0;
  bool is_grpc_response = false;
  std::vector<std::pair<std::string, std::string>> resp_headers;
};

struct H2ClientCtx {
  std::mutex mu;
  H2Session* session = nullptr;
  SocketId socket_id = 0;
  std::map<int32_t, H2ClientStream> streams;
  // closed streams are completed OUTSIDE mu (a completion can trigger a
  // retry that re-enters IssueH2Request -> mu: lock-order safety).
  std::vector<std::pair<H2ClientStream, uint32_t>> done_list;
  std::atomic<int> refs{1};

  ~H2ClientCtx() { delete session; }
};

void h2c_unref(H2ClientCtx* ctx) {
  if (ctx->refs.fetch_sub(1, std::memory_order_acq_rel) == 1) delete ctx;
}

void flush_client_locked(H2ClientCtx* ctx) {
  if (!ctx->session->has_output()) return;
  SocketUniquePtr sock;
  if (Socket::Address(ctx->socket_id, &sock) != 0) return;
  std::string bytes;
  ctx->session->TakeOutput(&bytes);
  if (!bytes.empty()) {
    IOBuf out;
    out.append(bytes);
    sock->Write(&out);
  }
}

// Runs with ctx->mu RELEASED (see struct comment).
namespace {
void fill_http_response(Controller* cntl, const H2ClientStream& st) {
  HttpHeaderExt& hr = cntl->http_response();
  hr.status_code = st.http_status;
  for (const auto& kv : st.resp_headers) {
    if (!kv.first.empty() && kv.first[0] != ':') hr.headers[kv.first] = kv.second;
    if (kv.first == "content-type") hr.content_type = kv.second;
  }
}
}  // namespace

void complete_stream(H2ClientCtx* ctx, H2ClientStream&& st, uint32_t error_code) {
  void* data = nullptr;
  if (session_lock(st.cid, &data) != 0) return;  // timed out already
  Controller* cntl = (Controller*)data;
  {
    SocketUniquePtr s;
    if (Socket::Address(ctx->socket_id, &s) == 0) s->remove_pending_session(st.cid);
  }
  if (error_code != 0) {
    cntl->SetFailed(EFAILEDSOCKET, "h2 stream reset, code " + std::to_string(error_code));
  } else if (st.grpc_status > 0) {
    cntl->SetFailed(st.grpc_status == 12 ? ENOMETHOD : EINTERNAL,
                    "grpc-status " + std::to_string(st.grpc_status) + ": " + st.grpc_message);
  } else if (st.http_status != 0 && st.http_status / 100 != 2) {
    fill_http_response(cntl, st);
    cntl->SetFailed(EHTTP, "h2 status " + std::to_string(st.http_status));
  } else {
    if (st.is_grpc_response && st.body.size() >= 5) st.body.pop_front(5);
    fill_http_response(cntl, st);
    if (cntl->call.response != nullptr) {
      cntl->call.response->clear();
      cntl->call.response->append(std::move(st.body));
    }
  }
  EndRPC(cntl, st.cid);
}

// ---- issue + parse hooks ----

int g_grpc_protocol_index = -1;
int g_h2c_protocol_index = -1;

bool own_client_ctx(Socket* sock) {
  return sock->protocol_ctx_owner == g_grpc_protocol_index ||
         sock->protocol_ctx_owner == g_h2c_protocol_index;
}

int IssueH2Request(Socket* sock, Controller* cntl, uint64_t cid, bool grpc) {
  if (sock->protocol_ctx != nullptr && !own_client_ctx(sock)) return -1;
  H2ClientCtx* ctx = (H2ClientCtx*)sock->protocol_ctx;
  if (ctx == nullptr) {
    ctx = new H2ClientCtx;
    ctx->socket_id = sock->id();
    H2Session::Callbacks cbs;
    cbs.on_header = [ctx](int32_t sid, const std::string& key, const std::string& val) {
      auto it = ctx->streams.find(sid);
      if (it == ctx->streams.end()) return;
      it->second.resp_headers.emplace_back(key, val);
      if (key == "grpc-status") it->second.grpc_status = atoi(val.c_str());
      else if (key == "grpc-message") it->second.grpc_message = val;
      else if (key == ":status") it->second.http_status = atoi(val.c_str());
      else if (key == "content-type" && val.rfind("application/grpc", 0) == 0)
        it->second.is_grpc_response = true;
    };
    cbs.on_data = [ctx](int32_t sid, const char* data, size_t n) {
      auto it = ctx->streams.find(sid);
      if (it != ctx->streams.end()) it->second.body.append(data, n);
    };
    cbs.on_end_stream = [ctx](int32_t sid) {
      auto it = ctx->streams.find(sid);
      if (it == ctx->streams.end()) return;
      ctx->done_list.emplace_back(std::move(it->second), 0);
      ctx->streams.erase(it);
    };
    cbs.on_rst = [ctx](int32_t sid, uint32_t error) {
      auto it = ctx->streams.find(sid);
      if (it == ctx->streams.end()) return;
      ctx->done_list.emplace_back(std::move(it->second), error != 0 ? error : 1);
      ctx->streams.erase(it);
    };
    ctx->session = new H2Session(/*server=*/false, cbs);
    sock->protocol_ctx = ctx;
    sock->protocol_ctx_deleter = [](void* p) { h2c_unref((H2ClientCtx*)p); };
    sock->protocol_ctx_owner = grpc ? g_grpc_protocol_index : g_h2c_protocol_index;
  }
  std::string payload = cntl->call.request_buf.to_string();
  std::string body;
  if (grpc) {
    // gRPC frame the payload (5-byte length prefix).
    char frame[5];
    frame[0] = 0;
    frame[1] = (char)(payload.size() >> 24);
    frame[2] = (char)(payload.size() >> 16);
    frame[3] = (char)(payload.size() >> 8);
    frame[4] = (char)payload.size();
    body.assign(frame, 5);
    body += payload;
  } else {
    body = payload;
  }

  const std::string& svc = cntl->call.service_name;
  const std::string& mn = cntl->call.method_name;
  std::string path = !mn.empty() && mn[0] == '/' ? mn
                     : !svc.empty() && svc[0] == '/' ? (mn.empty() ? svc : svc + "/" + mn)
                     : svc.empty() && !grpc          ? "/" + mn
                                                     : "/" + svc + "/" + mn;
  std::string authority = endpoint2str(cntl->remote_side());
  // Controller::http_request() overrides (same contract as the h1 client).
  const HttpHeaderExt* hx = cntl->has_http_request() ? &cntl->http_request() : nullptr;
  std::string verb = hx != nullptr && !hx->method.empty()
                         ? hx->method
                         : std::string(grpc || !payload.empty() ? "POST" : "GET");
  std::string ctype = grpc ? "application/grpc"
                     : hx != nullptr && !hx->content_type.empty()
                         ? hx->content_type
                         : "application/octet-stream";
  std::vector<hpack::Header> headers = {
      {":method", verb},
      {":scheme", "http"},
      {":path", path},
      {":authority", authority},
      {"content-type", ctype},
  };
  if (grpc) headers.push_back({"te", "trailers"});
  headers.push_back({"user-agent", grpc ? "brpc-amd-grpc/1.0" : "brpc-amd-h2/1.0"});
  if (hx != nullptr) {
    for (const auto& kv : hx->headers) headers.push_back({kv.first, kv.second});
  }
  std::lock_guard<std::mutex> lk(ctx->mu);
  int32_t stream_id = ctx->session->SubmitRequest(headers, body, /*end_stream=*/true);
  if (stream_id < 0) return -1;
  ctx->streams[stream_id].cid = cid;
  flush_client_locked(ctx);
  return 0;
}

ParseResult ParseGrpcClient(IOBuf* source, Socket* sock, bool /*eof*/) {
  if (sock->user() != nullptr) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  if (!own_client_ctx(sock)) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  H2ClientCtx* ctx = (H2ClientCtx*)sock->protocol_ctx;
  if (ctx == nullptr) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  std::string bytes = source->to_string();
  std::vector<std::pair<H2ClientStream, uint32_t>> done;
  {
    std::lock_guard<std::mutex> lk(ctx->mu);
    ssize_t consumed = ctx->session->Consume(bytes.data(), bytes.size());
    if (consumed < 0) return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
    source->pop_front((size_t)consumed);
    flush_client_locked(ctx);  // acks, window updates
    done.swap(ctx->done_list);
  }
  for (auto& d : done) complete_stream(ctx, std::move(d.first), d.second);
  // Responses complete above; no message object to dispatch.
  return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
}

}  // namespace

void RegisterGrpcClientProtocol() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    Protocol p;
    p.parse = ParseGrpcClient;
    p.issue_request = [](Socket* s, Controller* c, uint64_t cid) {
      return IssueH2Request(s, c, cid, /*grpc=*/true);
    };
    p.support_server = false;
    p.support_client = true;
    p.name = "grpc";
    g_grpc_protocol_index = RegisterProtocol(p);
    // Plain HTTP-semantics h2 client (parity: reference protocol "h2"):
    // same session machinery, no gRPC framing or trailers.
    Protocol h2;
    h2.parse = ParseGrpcClient;
    h2.issue_request = [](Socket* s, Controller* c, uint64_t cid) {
      return IssueH2Request(s, c, cid, /*grpc=*/false);
    };
    h2.support_server = false;
    h2.support_client = true;
    h2.name = "h2";
    g_h2c_protocol_index = RegisterProtocol(h2);
  });
}

}  // namespace policy
}  // namespace bam
