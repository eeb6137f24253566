// brpc_amd: HTTP/2 + gRPC server protocol.
// Parity: reference policy/http2_rpc_protocol.cpp + brpc/grpc.cpp (h2
// framing/HPACK + gRPC unary calls). Framing/HPACK ride the system
// libnghttp2 (same category as zlib for gzip — the h2 state machine is
// not a compute path); request routing, stream bookkeeping, gRPC 5-byte
// message framing, trailers and the socket integration are ours.
// The library is dlopened at runtime ("libnghttp2.so.14"), so builds
// never depend on link-time paths; if it is missing, the protocol simply
// does not register.
//
// gRPC interop is tested against the official grpc python client
// (tests/test_grpc.py).
#include <dlfcn.h>
#include <nghttp2/nghttp2.h>

#include <deque>
#include <map>
#include <mutex>

#include "base/logging.h"
#include "fiber/fiber.h"
#include "rpc/controller.h"
#include "rpc/policy/http_protocol.h"
#include "rpc/server.h"

namespace bam {
namespace policy {

namespace {

// ---- dlopened nghttp2 API ----
struct NgApi {
  int (*callbacks_new)(nghttp2_session_callbacks**);
  void (*callbacks_del)(nghttp2_session_callbacks*);
  void (*set_on_frame_recv)(nghttp2_session_callbacks*, nghttp2_on_frame_recv_callback);
  void (*set_on_header)(nghttp2_session_callbacks*, nghttp2_on_header_callback);
  void (*set_on_data_chunk)(nghttp2_session_callbacks*, nghttp2_on_data_chunk_recv_callback);
  void (*set_on_stream_close)(nghttp2_session_callbacks*, nghttp2_on_stream_close_callback);
  int (*server_new)(nghttp2_session**, const nghttp2_session_callbacks*, void*);
  void (*session_del)(nghttp2_session*);
  ssize_t (*mem_recv)(nghttp2_session*, const uint8_t*, size_t);
  ssize_t (*mem_send)(nghttp2_session*, const uint8_t**);
  int (*submit_settings)(nghttp2_session*, uint8_t, const nghttp2_settings_entry*, size_t);
  int (*submit_response)(nghttp2_session*, int32_t, const nghttp2_nv*, size_t,
                         const nghttp2_data_provider*);
  int (*submit_trailer)(nghttp2_session*, int32_t, const nghttp2_nv*, size_t);
  int (*session_want_write)(nghttp2_session*);
  bool ok = false;
};

NgApi& ng() {
  static NgApi api = [] {
    NgApi a;
    void* h = dlopen("libnghttp2.so.14", RTLD_NOW | RTLD_GLOBAL);
    if (h == nullptr) h = dlopen("libnghttp2.so", RTLD_NOW | RTLD_GLOBAL);
    if (h == nullptr) return a;
#define NG_SYM(field, name)                                   \
  *(void**)(&a.field) = dlsym(h, name);                       \
  if (a.field == nullptr) return a;
    NG_SYM(callbacks_new, "nghttp2_session_callbacks_new")
    NG_SYM(callbacks_del, "nghttp2_session_callbacks_del")
    NG_SYM(set_on_frame_recv, "nghttp2_session_callbacks_set_on_frame_recv_callback")
    NG_SYM(set_on_header, "nghttp2_session_callbacks_set_on_header_callback")
    NG_SYM(set_on_data_chunk, "nghttp2_session_callbacks_set_on_data_chunk_recv_callback")
    NG_SYM(set_on_stream_close, "nghttp2_session_callbacks_set_on_stream_close_callback")
    NG_SYM(server_new, "nghttp2_session_server_new")
    NG_SYM(session_del, "nghttp2_session_del")
    NG_SYM(mem_recv, "nghttp2_session_mem_recv")
    NG_SYM(mem_send, "nghttp2_session_mem_send")
    NG_SYM(submit_settings, "nghttp2_submit_settings")
    NG_SYM(submit_response, "nghttp2_submit_response")
    NG_SYM(submit_trailer, "nghttp2_submit_trailer")
    NG_SYM(session_want_write, "nghttp2_session_want_write")
#undef NG_SYM
    a.ok = true;
    return a;
  }();
  return api;
}

// ---- per-stream request state ----
struct H2Stream {
  std::map<std::string, std::string> headers;
  std::string path;
  std::string method;
  IOBuf body;
  bool is_grpc = false;
};

// ---- per-connection session context ----
struct H2Ctx {
  std::mutex mu;  // serializes ALL nghttp2_session access
  nghttp2_session* session = nullptr;
  SocketId socket_id = 0;
  std::map<int32_t, H2Stream> streams;
  std::deque<int32_t> completed;  // streams with END_STREAM, awaiting dispatch
  // 1 ref held by the socket + 1 per in-flight dispatch fiber.
  std::atomic<int> refs{1};

  ~H2Ctx() {
    if (session != nullptr) ng().session_del(session);
  }
};

void h2_ctx_unref(H2Ctx* ctx) {
  if (ctx->refs.fetch_sub(1, std::memory_order_acq_rel) == 1) delete ctx;
}

// response body holder handed to the data provider
struct H2ResponseBody {
  std::string data;
  size_t offset = 0;
  bool grpc = false;       // append grpc-status trailers
  int grpc_status = 0;
  std::string grpc_message;
};

nghttp2_nv make_nv(const char* name, const std::string& value) {
  nghttp2_nv nv;
  nv.name = (uint8_t*)name;
  nv.namelen = strlen(name);
  nv.value = (uint8_t*)value.data();
  nv.valuelen = value.size();
  nv.flags = NGHTTP2_NV_FLAG_NONE;
  return nv;
}

// pumps pending output bytes from the session to the socket. mu held.
void flush_session_locked(H2Ctx* ctx) {
  SocketUniquePtr sock;
  if (Socket::Address(ctx->socket_id, &sock) != 0) return;
  IOBuf out;
  for (;;) {
    const uint8_t* data = nullptr;
    ssize_t n = ng().mem_send(ctx->session, &data);
    if (n <= 0) break;
    out.append(data, (size_t)n);
  }
  if (!out.empty()) sock->Write(&out);
}

// ---- nghttp2 callbacks (session mutex held by the caller) ----

int on_header_cb(nghttp2_session*, const nghttp2_frame* frame, const uint8_t* name,
                 size_t namelen, const uint8_t* value, size_t valuelen, uint8_t,
                 void* user_data) {
  H2Ctx* ctx = (H2Ctx*)user_data;
  if (frame->hd.type != NGHTTP2_HEADERS) return 0;
  H2Stream& st = ctx->streams[frame->hd.stream_id];
  std::string key((const char*)name, namelen);
  std::string val((const char*)value, valuelen);
  if (key == ":path") st.path = val;
  else if (key == ":method") st.method = val;
  if (key == "content-type" && val.rfind("application/grpc", 0) == 0) st.is_grpc = true;
  st.headers[key] = val;
  return 0;
}

int on_data_chunk_cb(nghttp2_session*, uint8_t, int32_t stream_id, const uint8_t* data,
                     size_t len, void* user_data) {
  H2Ctx* ctx = (H2Ctx*)user_data;
  ctx->streams[stream_id].body.append(data, len);
  return 0;
}

int on_frame_recv_cb(nghttp2_session*, const nghttp2_frame* frame, void* user_data) {
  H2Ctx* ctx = (H2Ctx*)user_data;
  if ((frame->hd.type == NGHTTP2_HEADERS || frame->hd.type == NGHTTP2_DATA) &&
      (frame->hd.flags & NGHTTP2_FLAG_END_STREAM)) {
    ctx->completed.push_back(frame->hd.stream_id);
  }
  return 0;
}

int on_stream_close_cb(nghttp2_session*, int32_t stream_id, uint32_t, void* user_data) {
  H2Ctx* ctx = (H2Ctx*)user_data;
  ctx->streams.erase(stream_id);
  return 0;
}

// data provider read callback: streams the response body; for gRPC also
// submits the grpc-status trailers at EOF.
ssize_t response_read_cb(nghttp2_session* session, int32_t stream_id, uint8_t* buf,
                         size_t length, uint32_t* data_flags, nghttp2_data_source* source,
                         void* /*user_data*/) {
  H2ResponseBody* body = (H2ResponseBody*)source->ptr;
  size_t left = body->data.size() - body->offset;
  size_t n = left < length ? left : length;
  memcpy(buf, body->data.data() + body->offset, n);
  body->offset += n;
  if (body->offset >= body->data.size()) {
    *data_flags |= NGHTTP2_DATA_FLAG_EOF;
    if (body->grpc) {
      *data_flags |= NGHTTP2_DATA_FLAG_NO_END_STREAM;
      static thread_local std::string status_str, msg_str;
      status_str = std::to_string(body->grpc_status);
      msg_str = body->grpc_message;
      nghttp2_nv trailers[2] = {make_nv("grpc-status", status_str),
                                make_nv("grpc-message", msg_str)};
      ng().submit_trailer(session, stream_id, trailers,
                          body->grpc_message.empty() ? 1 : 2);
    }
    delete body;
    source->ptr = nullptr;
  }
  return (ssize_t)n;
}

// ---- request dispatch ----

struct H2DispatchArg {
  H2Ctx* ctx;
  int32_t stream_id;
  H2Stream stream;  // moved out of the session map
  Server* server;
};

void submit_h2_response_locked(H2Ctx* ctx, int32_t stream_id, int http_status,
                               const std::string& content_type, H2ResponseBody* body) {
  static thread_local std::string status_str;
  status_str = std::to_string(http_status);
  nghttp2_nv nvs[2] = {make_nv(":status", status_str), make_nv("content-type", content_type)};
  nghttp2_data_provider prd;
  prd.source.ptr = body;
  prd.read_callback = response_read_cb;
  ng().submit_response(ctx->session, stream_id, nvs, 2, &prd);
  flush_session_locked(ctx);
}

void h2_dispatch_fiber(void* raw) {
  H2DispatchArg* a = (H2DispatchArg*)raw;
  H2Stream& st = a->stream;
  // Route: /Service/Method (gRPC uses the same form, with package prefix).
  std::string path = st.path;
  if (!path.empty() && path[0] == '/') path = path.substr(1);
  size_t slash = path.find('/');
  std::string svc = slash == std::string::npos ? "" : path.substr(0, slash);
  std::string method = slash == std::string::npos ? path : path.substr(slash + 1);
  // gRPC service names may be package-qualified: try the last dot segment.
  const MethodFn* fn = a->server != nullptr ? a->server->FindMethod(svc, method) : nullptr;
  if (fn == nullptr && a->server != nullptr) {
    size_t dot = svc.find_last_of('.');
    if (dot != std::string::npos) fn = a->server->FindMethod(svc.substr(dot + 1), method);
  }

  H2ResponseBody* body = new H2ResponseBody;
  body->grpc = st.is_grpc;
  int http_status = 200;
  std::string content_type = st.is_grpc ? "application/grpc" : "application/octet-stream";

  if (fn == nullptr) {
    if (st.is_grpc && st.path == "/grpc.health.v1.Health/Check") {
      // builtin gRPC health service (parity: reference grpc_health_check):
      // HealthCheckResponse{status: SERVING} = field 1 varint 1
      std::string payload("\x08\x01", 2);
      char frame[5] = {0, 0, 0, 0, (char)payload.size()};
      body->data.assign(frame, 5);
      body->data += payload;
      body->grpc_status = 0;
    } else if (st.is_grpc) {
      body->grpc_status = 12;  // UNIMPLEMENTED
      body->grpc_message = "unknown method " + st.path;
    } else {
      // fall back to builtin pages over h2
      HttpRequest req;
      req.path = st.path;
      req.method = st.method;
      HttpResponse resp;
      if (DispatchBuiltinService(a->server, req, &resp)) {
        http_status = resp.status;
        content_type = resp.content_type;
        body->data = resp.body.to_string();
      } else {
        http_status = 404;
        body->data = "no such method\n";
      }
    }
  } else {
    IOBuf request_payload;
    if (st.is_grpc) {
      // strip the 5-byte gRPC message frame (compressed flag + u32 len)
      if (st.body.size() >= 5) {
        st.body.pop_front(5);
      }
      request_payload.swap(st.body);
    } else {
      request_payload.swap(st.body);
    }
    Controller cntl;
    cntl.server_ = a->server;
    IOBuf response_payload;
    std::atomic<bool> done_flag{false};
    Closure* done = NewCallback([&done_flag] { done_flag.store(true); });
    (*fn)(&cntl, request_payload, &response_payload, done);
    for (int i = 0; i < 300000 && !done_flag.load(std::memory_order_acquire); ++i) {
      fiber_usleep(100);
    }
    if (cntl.Failed()) {
      if (st.is_grpc) {
        body->grpc_status = 13;  // INTERNAL
        body->grpc_message = cntl.ErrorText();
      } else {
        http_status = 500;
        body->data = cntl.ErrorText();
      }
    } else {
      std::string payload = response_payload.to_string();
      if (st.is_grpc) {
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
    }
  }
  {
    std::lock_guard<std::mutex> lk(a->ctx->mu);
    if (a->ctx->session != nullptr) {
      submit_h2_response_locked(a->ctx, a->stream_id, http_status, content_type, body);
    } else {
      delete body;
    }
  }
  h2_ctx_unref(a->ctx);
  delete a;
}

// ---- protocol hooks ----

struct H2PumpMessage : public InputMessageBase {
  // completed streams snapshot to dispatch
  std::vector<H2DispatchArg*> dispatches;
};

const char kPreface[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";  // 24 bytes

int g_h2_protocol_index = -1;

ParseResult ParseH2(IOBuf* source, Socket* sock, bool /*eof*/) {
  if (sock->user() == nullptr) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  if (sock->protocol_ctx != nullptr && sock->protocol_ctx_owner != g_h2_protocol_index)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);  // another protocol's connection
  H2Ctx* ctx = (H2Ctx*)sock->protocol_ctx;
  if (ctx == nullptr) {
    // new connection: require the client preface
    if (source->size() < 24) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
    char aux[24];
    const char* h = (const char*)source->fetch(aux, 24);
    if (memcmp(h, kPreface, 24) != 0) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
    if (!ng().ok) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
    ctx = new H2Ctx;
    ctx->socket_id = sock->id();
    nghttp2_session_callbacks* cbs = nullptr;
    ng().callbacks_new(&cbs);
    ng().set_on_frame_recv(cbs, on_frame_recv_cb);
    ng().set_on_header(cbs, on_header_cb);
    ng().set_on_data_chunk(cbs, on_data_chunk_cb);
    ng().set_on_stream_close(cbs, on_stream_close_cb);
    ng().server_new(&ctx->session, cbs, ctx);
    ng().callbacks_del(cbs);
    ng().submit_settings(ctx->session, NGHTTP2_FLAG_NONE, nullptr, 0);
    sock->protocol_ctx = ctx;
    sock->protocol_ctx_deleter = [](void* p) { h2_ctx_unref((H2Ctx*)p); };
    sock->protocol_ctx_owner = g_h2_protocol_index;
  }
  // Feed everything we have into the session.
  std::string bytes = source->to_string();
  H2PumpMessage* msg = new H2PumpMessage;
  Server* server = (Server*)sock->user();
  {
    std::lock_guard<std::mutex> lk(ctx->mu);
    ssize_t consumed = ng().mem_recv(ctx->session, (const uint8_t*)bytes.data(), bytes.size());
    if (consumed < 0) {
      delete msg;
      return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
    }
    source->pop_front((size_t)consumed);
    // collect completed requests
    while (!ctx->completed.empty()) {
      int32_t sid = ctx->completed.front();
      ctx->completed.pop_front();
      auto it = ctx->streams.find(sid);
      if (it == ctx->streams.end()) continue;
      ctx->refs.fetch_add(1, std::memory_order_acq_rel);
      H2DispatchArg* a = new H2DispatchArg{ctx, sid, std::move(it->second), server};
      msg->dispatches.push_back(a);
    }
    flush_session_locked(ctx);  // settings ack, window updates, ...
  }
  if (msg->dispatches.empty()) {
    delete msg;
    // We consumed bytes; tell the messenger to keep polling this protocol.
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  }
  return ParseResult::make_ok(msg);
}

void ProcessH2(InputMessageBase* msg_base) {
  H2PumpMessage* msg = (H2PumpMessage*)msg_base;
  for (H2DispatchArg* a : msg->dispatches) {
    fiber_t th;
    if (fiber_start_background(&th, h2_dispatch_fiber, a) != 0) h2_dispatch_fiber(a);
  }
  delete msg;
}

}  // namespace

void RegisterH2Protocol() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    if (!ng().ok) {
      LOG(WARNING) << "libnghttp2 unavailable; h2/gRPC protocol disabled";
      return;
    }
    Protocol p;
    p.parse = ParseH2;
    p.process_request = ProcessH2;
    p.process_response = nullptr;
    p.support_server = true;
    p.support_client = false;
    p.name = "h2";
    g_h2_protocol_index = RegisterProtocol(p);
  });
}

}  // namespace policy
}  // namespace bam
