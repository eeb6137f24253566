// brpc_amd: HTTP/2 / gRPC CLIENT protocol (channel protocol "grpc").
// Parity: reference h2/gRPC client paths in policy/http2_rpc_protocol.cpp.
// One nghttp2 client session per connection (socket protocol_ctx);
// requests map stream_id -> correlation session; responses complete on
// stream close with grpc-status trailer handling. Interop-tested against
// the official grpc python SERVER (tests/test_grpc.py).
#include <dlfcn.h>
#include <nghttp2/nghttp2.h>

#include <map>
#include <mutex>

#include "base/logging.h"
#include "fiber/session.h"
#include "rpc/channel.h"
#include "rpc/controller.h"
#include "rpc/server.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

namespace policy {

namespace {

struct NgcApi {
  int (*callbacks_new)(nghttp2_session_callbacks**);
  void (*callbacks_del)(nghttp2_session_callbacks*);
  void (*set_on_frame_recv)(nghttp2_session_callbacks*, nghttp2_on_frame_recv_callback);
  void (*set_on_header)(nghttp2_session_callbacks*, nghttp2_on_header_callback);
  void (*set_on_data_chunk)(nghttp2_session_callbacks*, nghttp2_on_data_chunk_recv_callback);
  void (*set_on_stream_close)(nghttp2_session_callbacks*, nghttp2_on_stream_close_callback);
  int (*client_new)(nghttp2_session**, const nghttp2_session_callbacks*, void*);
  void (*session_del)(nghttp2_session*);
  ssize_t (*mem_recv)(nghttp2_session*, const uint8_t*, size_t);
  ssize_t (*mem_send)(nghttp2_session*, const uint8_t**);
  int (*submit_settings)(nghttp2_session*, uint8_t, const nghttp2_settings_entry*, size_t);
  int32_t (*submit_request)(nghttp2_session*, const nghttp2_priority_spec*,
                            const nghttp2_nv*, size_t, const nghttp2_data_provider*, void*);
  bool ok = false;
};

NgcApi& ngc() {
  static NgcApi api = [] {
    NgcApi a;
    void* h = dlopen("libnghttp2.so.14", RTLD_NOW | RTLD_GLOBAL);
    if (h == nullptr) h = dlopen("libnghttp2.so", RTLD_NOW | RTLD_GLOBAL);
    if (h == nullptr) return a;
#define NGC_SYM(field, name)            \
  *(void**)(&a.field) = dlsym(h, name); \
  if (a.field == nullptr) return a;
    NGC_SYM(callbacks_new, "nghttp2_session_callbacks_new")
    NGC_SYM(callbacks_del, "nghttp2_session_callbacks_del")
    NGC_SYM(set_on_frame_recv, "nghttp2_session_callbacks_set_on_frame_recv_callback")
    NGC_SYM(set_on_header, "nghttp2_session_callbacks_set_on_header_callback")
    NGC_SYM(set_on_data_chunk, "nghttp2_session_callbacks_set_on_data_chunk_recv_callback")
    NGC_SYM(set_on_stream_close, "nghttp2_session_callbacks_set_on_stream_close_callback")
    NGC_SYM(client_new, "nghttp2_session_client_new")
    NGC_SYM(session_del, "nghttp2_session_del")
    NGC_SYM(mem_recv, "nghttp2_session_mem_recv")
    NGC_SYM(mem_send, "nghttp2_session_mem_send")
    NGC_SYM(submit_settings, "nghttp2_submit_settings")
    NGC_SYM(submit_request, "nghttp2_submit_request")
#undef NGC_SYM
    a.ok = true;
    return a;
  }();
  return api;
}

struct H2ClientStream {
  uint64_t cid = 0;
  IOBuf body;
  int grpc_status = -1;
  std::string grpc_message;
  int http_status = 0;
  bool is_grpc_response = false;
};

struct H2ClientCtx {
  std::mutex mu;
  nghttp2_session* session = nullptr;
  SocketId socket_id = 0;
  std::map<int32_t, H2ClientStream> streams;
  // closed streams are completed OUTSIDE mu (a completion can trigger a
  // retry that re-enters IssueGrpcRequest -> mu: lock-order safety).
  std::vector<std::pair<H2ClientStream, uint32_t>> done_list;
  std::atomic<int> refs{1};

  ~H2ClientCtx() {
    if (session != nullptr) ngc().session_del(session);
  }
};

void h2c_unref(H2ClientCtx* ctx) {
  if (ctx->refs.fetch_sub(1, std::memory_order_acq_rel) == 1) delete ctx;
}

struct H2ClientBody {
  std::string data;
  size_t offset = 0;
};

nghttp2_nv cnv(const char* name, const std::string& value) {
  nghttp2_nv nv;
  nv.name = (uint8_t*)name;
  nv.namelen = strlen(name);
  nv.value = (uint8_t*)value.data();
  nv.valuelen = value.size();
  nv.flags = NGHTTP2_NV_FLAG_NONE;
  return nv;
}

void flush_client_locked(H2ClientCtx* ctx) {
  SocketUniquePtr sock;
  if (Socket::Address(ctx->socket_id, &sock) != 0) return;
  IOBuf out;
  for (;;) {
    const uint8_t* data = nullptr;
    ssize_t n = ngc().mem_send(ctx->session, &data);
    if (n <= 0) break;
    out.append(data, (size_t)n);
  }
  if (!out.empty()) sock->Write(&out);
}

// ---- callbacks (ctx->mu held by the pump) ----

int c_on_header(nghttp2_session*, const nghttp2_frame* frame, const uint8_t* name,
                size_t namelen, const uint8_t* value, size_t valuelen, uint8_t,
                void* user_data) {
  H2ClientCtx* ctx = (H2ClientCtx*)user_data;
  auto it = ctx->streams.find(frame->hd.stream_id);
  if (it == ctx->streams.end()) return 0;
  std::string key((const char*)name, namelen);
  std::string val((const char*)value, valuelen);
  if (key == "grpc-status") it->second.grpc_status = atoi(val.c_str());
  else if (key == "grpc-message") it->second.grpc_message = val;
  else if (key == ":status") it->second.http_status = atoi(val.c_str());
  else if (key == "content-type" && val.rfind("application/grpc", 0) == 0)
    it->second.is_grpc_response = true;
  return 0;
}

int c_on_data(nghttp2_session*, uint8_t, int32_t stream_id, const uint8_t* data, size_t len,
              void* user_data) {
  H2ClientCtx* ctx = (H2ClientCtx*)user_data;
  auto it = ctx->streams.find(stream_id);
  if (it != ctx->streams.end()) it->second.body.append(data, len);
  return 0;
}

int c_on_stream_close(nghttp2_session*, int32_t stream_id, uint32_t error_code,
                      void* user_data) {
  H2ClientCtx* ctx = (H2ClientCtx*)user_data;
  auto it = ctx->streams.find(stream_id);
  if (it == ctx->streams.end()) return 0;
  ctx->done_list.emplace_back(std::move(it->second), error_code);
  ctx->streams.erase(it);
  return 0;
}

// Runs with ctx->mu RELEASED (see struct comment).
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
  } else if (st.http_status != 0 && st.http_status != 200) {
    cntl->SetFailed(EHTTP, "h2 status " + std::to_string(st.http_status));
  } else {
    if (st.is_grpc_response && st.body.size() >= 5) st.body.pop_front(5);
    if (cntl->call.response != nullptr) {
      cntl->call.response->clear();
      cntl->call.response->append(std::move(st.body));
    }
  }
  EndRPC(cntl, st.cid);
}

ssize_t c_body_read(nghttp2_session*, int32_t, uint8_t* buf, size_t length,
                    uint32_t* data_flags, nghttp2_data_source* source, void*) {
  H2ClientBody* body = (H2ClientBody*)source->ptr;
  size_t left = body->data.size() - body->offset;
  size_t n = left < length ? left : length;
  memcpy(buf, body->data.data() + body->offset, n);
  body->offset += n;
  if (body->offset >= body->data.size()) {
    *data_flags |= NGHTTP2_DATA_FLAG_EOF;
    delete body;
    source->ptr = nullptr;
  }
  return (ssize_t)n;
}

// ---- issue + parse hooks ----

int g_grpc_protocol_index = -1;
int g_h2c_protocol_index = -1;

bool own_client_ctx(Socket* sock) {
  return sock->protocol_ctx_owner == g_grpc_protocol_index ||
         sock->protocol_ctx_owner == g_h2c_protocol_index;
}

int IssueH2Request(Socket* sock, Controller* cntl, uint64_t cid, bool grpc) {
  if (!ngc().ok) return -1;
  if (sock->protocol_ctx != nullptr && !own_client_ctx(sock)) return -1;
  H2ClientCtx* ctx = (H2ClientCtx*)sock->protocol_ctx;
  if (ctx == nullptr) {
    ctx = new H2ClientCtx;
    ctx->socket_id = sock->id();
    nghttp2_session_callbacks* cbs = nullptr;
    ngc().callbacks_new(&cbs);
    ngc().set_on_header(cbs, c_on_header);
    ngc().set_on_data_chunk(cbs, c_on_data);
    ngc().set_on_stream_close(cbs, c_on_stream_close);
    ngc().client_new(&ctx->session, cbs, ctx);
    ngc().callbacks_del(cbs);
    ngc().submit_settings(ctx->session, NGHTTP2_FLAG_NONE, nullptr, 0);
    sock->protocol_ctx = ctx;
    sock->protocol_ctx_deleter = [](void* p) { h2c_unref((H2ClientCtx*)p); };
    sock->protocol_ctx_owner = grpc ? g_grpc_protocol_index : g_h2c_protocol_index;
  }
  std::string payload = cntl->call.request_buf.to_string();
  H2ClientBody* body = new H2ClientBody;
  if (grpc) {
    // gRPC frame the payload (5-byte length prefix).
    char frame[5];
    frame[0] = 0;
    frame[1] = (char)(payload.size() >> 24);
    frame[2] = (char)(payload.size() >> 16);
    frame[3] = (char)(payload.size() >> 8);
    frame[4] = (char)payload.size();
    body->data.assign(frame, 5);
    body->data += payload;
  } else {
    body->data = payload;
  }

  const std::string& svc = cntl->call.service_name;
  const std::string& mn = cntl->call.method_name;
  std::string path = !mn.empty() && mn[0] == '/' ? mn
                     : !svc.empty() && svc[0] == '/' ? svc + "/" + mn
                     : svc.empty() && !grpc          ? "/" + mn
                                                     : "/" + svc + "/" + mn;
  std::string authority = endpoint2str(cntl->remote_side());
  // NOTE: every value must outlive submit_request — named locals, never
  // temporaries (nghttp2_nv holds raw pointers).
  std::string v_method = grpc || !payload.empty() ? "POST" : "GET", v_scheme = "http",
              v_ct = grpc ? "application/grpc" : "application/octet-stream",
              v_te = "trailers", v_ua = grpc ? "brpc-amd-grpc/1.0" : "brpc-amd-h2/1.0";
  nghttp2_nv nvs[7] = {
      cnv(":method", v_method), cnv(":scheme", v_scheme), cnv(":path", path),
      cnv(":authority", authority), cnv("content-type", v_ct), cnv("te", v_te),
      cnv("user-agent", v_ua),
  };
  // plain h2 omits "te: trailers" (gRPC-specific).
  const size_t nnv = grpc ? 7 : 6;
  if (!grpc) nvs[5] = nvs[6];  // drop te, keep user-agent
  nghttp2_data_provider prd;
  prd.source.ptr = body;
  prd.read_callback = c_body_read;
  std::lock_guard<std::mutex> lk(ctx->mu);
  int32_t stream_id =
      ngc().submit_request(ctx->session, nullptr, nvs, nnv, &prd, nullptr);
  if (stream_id < 0) {
    delete body;
    return -1;
  }
  ctx->streams[stream_id].cid = cid;
  flush_client_locked(ctx);
  return 0;
}

struct H2ClientPump : public InputMessageBase {};

ParseResult ParseGrpcClient(IOBuf* source, Socket* sock, bool /*eof*/) {
  if (sock->user() != nullptr) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  if (!own_client_ctx(sock)) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  H2ClientCtx* ctx = (H2ClientCtx*)sock->protocol_ctx;
  if (ctx == nullptr) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  std::string bytes = source->to_string();
  std::vector<std::pair<H2ClientStream, uint32_t>> done;
  {
    std::lock_guard<std::mutex> lk(ctx->mu);
    ssize_t consumed = ngc().mem_recv(ctx->session, (const uint8_t*)bytes.data(), bytes.size());
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
    if (!ngc().ok) {
      LOG(WARNING) << "libnghttp2 unavailable; grpc client protocol disabled";
      return;
    }
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
    // same nghttp2 session machinery, no gRPC framing or trailers.
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
