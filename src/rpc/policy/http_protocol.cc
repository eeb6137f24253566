// brpc_amd: HTTP/1.1 server-side protocol.
// Parity: reference policy/http_rpc_protocol.cpp + details/http_parser —
// serves (a) the builtin service pages (/status /vars /flags /health
// /connections /index ... see rpc/builtin/) and (b) RPC-over-HTTP:
// POST/GET /ServiceName/MethodName with the body as request payload.
// Hand-written incremental parser (request line + headers + content-length
// body); keep-alive + pipelining via the standard parse loop.
#include "rpc/policy/http_protocol.h"

#include <unistd.h>

#include <algorithm>
#include <mutex>

#include "base/flags.h"
#include "base/flat_map.h"
#include "base/logging.h"
#include "base/time.h"
#include "rpc/controller.h"
#include "rpc/server.h"
#include "rpc/wire.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

namespace policy {

// Parity: reference FLAGS_max_body_size (http_message.cpp) — a declared
// Content-Length (or chunked total) beyond this is rejected outright so a
// hostile peer cannot grow read_buf_ without bound.
BAM_DEFINE_int64(http_max_body_size, 64 << 20,
                 "Reject HTTP messages whose body exceeds this many bytes");

namespace {

// Strict non-negative decimal parse for Content-Length / chunk totals.
// Returns false on empty/garbage/negative/overflow/over-limit values.
bool parse_body_size(const char* s, size_t* out) {
  if (s == nullptr || *s == '\0') return false;
  uint64_t v = 0;
  const char* p = s;
  for (; *p != '\0' && *p != '\r' && *p != ' '; ++p) {
    if (*p < '0' || *p > '9') return false;
    if (v > (UINT64_MAX - 9) / 10) return false;
    v = v * 10 + (uint64_t)(*p - '0');
  }
  if (p == s) return false;
  if ((int64_t)v > FLAG_http_max_body_size) return false;
  *out = (size_t)v;
  return true;
}

struct HttpMessage : public InputMessageBase {
  HttpRequest req;
};

// Finds "\r\n\r\n" within the first `limit` bytes of buf. Returns offset of
// header end (past the blank line) or npos.
size_t find_header_end(const std::string& head) {
  size_t p = head.find("\r\n\r\n");
  return p == std::string::npos ? std::string::npos : p + 4;
}

bool parse_headers(const std::string& head, HttpRequest* out) {
  size_t line_end = head.find("\r\n");
  if (line_end == std::string::npos) return false;
  // request line: METHOD SP PATH SP VERSION
  const std::string line = head.substr(0, line_end);
  size_t sp1 = line.find(' ');
  size_t sp2 = line.rfind(' ');
  if (sp1 == std::string::npos || sp2 == sp1) return false;
  out->method = line.substr(0, sp1);
  std::string uri = line.substr(sp1 + 1, sp2 - sp1 - 1);
  size_t q = uri.find('?');
  if (q == std::string::npos) {
    out->path = uri;
  } else {
    out->path = uri.substr(0, q);
    std::string query = uri.substr(q + 1);
    size_t pos = 0;
    while (pos < query.size()) {
      size_t amp = query.find('&', pos);
      std::string kv = query.substr(pos, amp == std::string::npos ? amp : amp - pos);
      size_t eq = kv.find('=');
      if (eq == std::string::npos) {
        out->query[kv] = "";
      } else {
        out->query[kv.substr(0, eq)] = kv.substr(eq + 1);
      }
      if (amp == std::string::npos) break;
      pos = amp + 1;
    }
  }
  size_t pos = line_end + 2;
  while (pos < head.size()) {
    size_t eol = head.find("\r\n", pos);
    if (eol == std::string::npos || eol == pos) break;
    std::string hline = head.substr(pos, eol - pos);
    size_t colon = hline.find(':');
    if (colon != std::string::npos) {
      std::string key = hline.substr(0, colon);
      size_t vstart = hline.find_first_not_of(' ', colon + 1);
      std::string val = vstart == std::string::npos ? "" : hline.substr(vstart);
      // lower-case keys for lookup
      std::transform(key.begin(), key.end(), key.begin(), ::tolower);
      out->headers[key] = val;
    }
    pos = eol + 2;
  }
  return true;
}

// ---- client side (HTTP/1.1 responses, FIFO-correlated) ----
// Parity: reference http_rpc_protocol.cpp client half. Supports
// Content-Length and chunked transfer-coding.

struct HttpClientMessage : public InputMessageBase {
  int status = 0;
  std::string reason;
  std::map<std::string, std::string> headers;
  IOBuf body;
};

// ---- progressive response state (Controller::response_read_progressively,
// parity: reference ProgressiveReader / socket.h:662) ----
// Lives as the client socket's protocol_ctx while a progressive body is
// being streamed; body bytes are delivered to the reader as they arrive
// instead of buffering (the 1 MB+ payload path).
const std::string CRLF("\r\n");

struct HttpProgressiveCtx {
  bool active = false;
  bool chunked = false;
  size_t remaining = 0;    // content-length mode
  int64_t chunk_left = -1; // chunked mode: -1 = expect a chunk-size line
  uint64_t cid = 0;
};

int g_http_protocol_index = -1;

// Feeds available body bytes to the progressive reader. Returns true when
// the body completed (RPC ended), false when more bytes are needed.
bool progressive_pump(IOBuf* source, Socket* sock, HttpProgressiveCtx* ctx) {
  for (;;) {
    IOBuf chunk;
    bool done = false;
    if (!ctx->chunked) {
      size_t take = std::min(ctx->remaining, source->size());
      if (take == 0 && ctx->remaining > 0) return false;
      source->cutn(&chunk, take);
      ctx->remaining -= take;
      done = ctx->remaining == 0;
    } else {
      if (ctx->chunk_left < 0) {
        // need a "<hex>CRLF" size line
        IOBuf line;
        if (source->cut_until(&line, CRLF) != 0) return false;
        std::string l = line.to_string();
        char* endp = nullptr;
        unsigned long long v = strtoull(l.c_str(), &endp, 16);
        if (endp == l.c_str() || v > (unsigned long long)FLAG_http_max_body_size) {
          // malformed: fail the call below via a zero-length done
          ctx->chunk_left = 0;
          v = 0;
        }
        ctx->chunk_left = (int64_t)v;
        if (ctx->chunk_left == 0) {
          // consume trailing CRLF (trailer-less)
          IOBuf fin;
          if (source->cut_until(&fin, CRLF) != 0) return false;
          done = true;
        }
      }
      if (!done && ctx->chunk_left > 0) {
        size_t take = std::min((size_t)ctx->chunk_left, source->size());
        if (take == 0) return false;
        source->cutn(&chunk, take);
        ctx->chunk_left -= (int64_t)take;
        if (ctx->chunk_left == 0) {
          IOBuf crlf;
          if (source->size() >= 2) {
            source->cutn(&crlf, 2);
            ctx->chunk_left = -1;
          } else {
            // deliver this chunk now; CRLF on the next read
            ctx->chunk_left = 0;
            void* data = nullptr;
            if (session_lock(ctx->cid, &data) == 0) {
              Controller* cntl = (Controller*)data;
              if (cntl->progressive_reader()) cntl->progressive_reader()(chunk, false);
              session_unlock(ctx->cid);
            }
            return false;
          }
        }
      } else if (!done && ctx->chunk_left == 0) {
        // pending CRLF from a split chunk boundary
        if (source->size() < 2) return false;
        IOBuf crlf;
        source->cutn(&crlf, 2);
        ctx->chunk_left = -1;
        continue;
      }
    }
    // deliver
    void* data = nullptr;
    if (session_lock(ctx->cid, &data) == 0) {
      Controller* cntl = (Controller*)data;
      if (cntl->progressive_reader()) cntl->progressive_reader()(chunk, done);
      if (done) {
        sock->remove_pending_session(ctx->cid);
        sock->pop_pipeline();
        ctx->active = false;
        EndRPC(cntl, ctx->cid);
        return true;
      }
      session_unlock(ctx->cid);
    } else if (done) {
      // call timed out mid-body: drop the rest quietly
      sock->pop_pipeline();
      ctx->active = false;
      return true;
    }
    if (done) return true;
    if (!ctx->chunked && ctx->remaining == 0) return true;
    if (source->empty()) return false;
  }
}

ParseResult ParseHttpResponse(IOBuf* source, Socket* sock, bool eof) {
  // Resume an in-flight progressive body first.
  if (sock->protocol_ctx != nullptr && sock->protocol_ctx_owner == g_http_protocol_index) {
    HttpProgressiveCtx* pctx = (HttpProgressiveCtx*)sock->protocol_ctx;
    if (pctx->active) {
      if (!progressive_pump(source, sock, pctx))
        return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
      if (source->empty()) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
      // fall through: the next pipelined response is already buffered
    }
  }
  char probe[8];
  size_t n = std::min<size_t>(source->size(), 8);
  if (n < 5) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* p = (const char*)source->fetch(probe, n);
  if (memcmp(p, "HTTP/", 5) != 0) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  std::string head;
  size_t scan = std::min<size_t>(source->size(), 64 << 10);
  source->copy_to(&head, scan, 0);
  size_t hend = find_header_end(head);
  if (hend == std::string::npos) {
    if (scan >= (64 << 10)) return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  }
  // status line: HTTP/1.1 200 OK
  size_t sp1 = head.find(' ');
  size_t eol = head.find("\r\n");
  if (sp1 == std::string::npos || eol == std::string::npos || sp1 > eol)
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  int status = atoi(head.c_str() + sp1 + 1);
  std::map<std::string, std::string> headers;
  size_t line = eol + 2;
  while (line + 2 <= hend) {
    size_t le = head.find("\r\n", line);
    if (le == std::string::npos || le >= hend) break;
    std::string l = head.substr(line, le - line);
    line = le + 2;
    if (l.empty()) break;
    size_t c = l.find(':');
    if (c == std::string::npos) continue;
    std::string k = l.substr(0, c);
    for (char& ch : k) ch = (char)tolower((unsigned char)ch);
    size_t v0 = l.find_first_not_of(' ', c + 1);
    headers[k] = v0 == std::string::npos ? "" : l.substr(v0);
  }
  // Progressive delivery: if the head-of-pipeline call asked for it,
  // stream body bytes to its reader instead of buffering.
  auto te_probe = headers.find("transfer-encoding");
  const bool is_chunked =
      te_probe != headers.end() && te_probe->second.find("chunked") != std::string::npos;
  if (status == 200) {
    uint64_t cid = sock->peek_pipeline();
    void* data = nullptr;
    if (cid != 0 && session_lock(cid, &data) == 0) {
      Controller* cntl = (Controller*)data;
      const bool want = (bool)cntl->progressive_reader();
      session_unlock(cid);
      if (want) {
        HttpProgressiveCtx* pctx = nullptr;
        if (sock->protocol_ctx != nullptr &&
            sock->protocol_ctx_owner == g_http_protocol_index) {
          pctx = (HttpProgressiveCtx*)sock->protocol_ctx;
        } else if (sock->protocol_ctx == nullptr) {
          pctx = new HttpProgressiveCtx;
          sock->protocol_ctx = pctx;
          sock->protocol_ctx_deleter = [](void* q) { delete (HttpProgressiveCtx*)q; };
          sock->protocol_ctx_owner = g_http_protocol_index;
        }
        if (pctx != nullptr) {
          source->pop_front(hend);  // headers consumed; body streams out
          pctx->active = true;
          pctx->cid = cid;
          pctx->chunked = is_chunked;
          pctx->chunk_left = -1;
          pctx->remaining = 0;
          if (!is_chunked) {
            auto cl = headers.find("content-length");
            if (cl != headers.end() &&
                !parse_body_size(cl->second.c_str(), &pctx->remaining)) {
              pctx->active = false;
              return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
            }
          }
          if (!progressive_pump(source, sock, pctx) || source->empty())
            return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
          return ParseHttpResponse(source, sock, eof);  // next pipelined response
        }
      }
    }
  }
  IOBuf body;
  size_t consumed = hend;
  auto te = headers.find("transfer-encoding");
  if (te != headers.end() && te->second.find("chunked") != std::string::npos) {
    // Decode chunked coding; needs the terminating 0-chunk in the buffer.
    std::string all;
    source->copy_to(&all, (size_t)-1, 0);
    size_t pos = hend;
    for (;;) {
      size_t le = all.find("\r\n", pos);
      if (le == std::string::npos) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
      char* cend = nullptr;
      unsigned long long chunk_ull = strtoull(all.c_str() + pos, &cend, 16);
      if (cend == all.c_str() + pos || all[pos] == '-' ||
          chunk_ull > (unsigned long long)FLAG_http_max_body_size ||
          body.size() + chunk_ull > (size_t)FLAG_http_max_body_size) {
        return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
      }
      size_t chunk_len = (size_t)chunk_ull;
      pos = le + 2;
      if (chunk_len == 0) {
        // trailer section ends with CRLF
        size_t fin = all.find("\r\n", pos);
        if (fin == std::string::npos) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
        consumed = fin + 2;
        break;
      }
      if (all.size() < pos + chunk_len + 2)
        return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
      body.append(all.data() + pos, chunk_len);
      pos += chunk_len + 2;
    }
    source->pop_front(consumed);
  } else {
    size_t content_len = 0;
    auto cl = headers.find("content-length");
    if (cl != headers.end() && !parse_body_size(cl->second.c_str(), &content_len))
      return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
    if (source->size() < hend + content_len)
      return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
    source->pop_front(hend);
    source->cutn(&body, content_len);
  }
  HttpClientMessage* msg = new HttpClientMessage;
  msg->status = status;
  msg->headers.swap(headers);
  msg->body.swap(body);
  return ParseResult::make_ok(msg);
}

void ProcessHttpResponse(InputMessageBase* mb) {
  HttpClientMessage* msg = (HttpClientMessage*)mb;
  SocketUniquePtr sock;
  SessionId cid = 0;
  if (Socket::Address(msg->socket_id, &sock) == 0) cid = sock->pop_pipeline();
  void* data = nullptr;
  if (cid == 0 || session_lock(cid, &data) != 0) {
    delete msg;
    return;
  }
  Controller* cntl = (Controller*)data;
  sock->remove_pending_session(cid);
  HttpHeaderExt& hr = cntl->http_response();
  hr.status_code = msg->status;
  for (const auto& kv : msg->headers) hr.headers[kv.first] = kv.second;
  auto ctit = msg->headers.find("content-type");
  if (ctit != msg->headers.end()) hr.content_type = ctit->second;
  if (msg->status / 100 != 2) {  // any 2xx is success (204, 202…)
    cntl->SetFailed(EHTTP, "HTTP status " + std::to_string(msg->status) + ": " +
                               msg->body.to_string().substr(0, 200));
    // keep the error page readable (≙ reference: body stays accessible
    // via response_attachment on HTTP errors)
    cntl->response_attachment().clear();
    cntl->response_attachment().swap(msg->body);
  } else if (cntl->call.response != nullptr) {
    cntl->call.response->clear();
    cntl->call.response->swap(msg->body);
  }
  delete msg;
  EndRPC(cntl, cid);
}

void PackHttp1Request(IOBuf* out, Controller* cntl, uint64_t /*fifo-correlated*/) {
  const std::string& svc = cntl->call.service_name;
  const std::string& m = cntl->call.method_name;
  // Absolute-path methods arrive whole in svc (split_full_method).
  std::string path = !m.empty() && m[0] == '/' ? m
                     : !svc.empty() && svc[0] == '/' ? (m.empty() ? svc : svc + "/" + m)
                     : svc.empty()                   ? "/" + m
                                                     : "/" + svc + "/" + m;
  const IOBuf& body = cntl->call.request_buf;
  // Controller::http_request() overrides (≙ reference HttpHeader on the
  // client side): custom verb, content-type and extra headers.
  const HttpHeaderExt* hx = cntl->has_http_request() ? &cntl->http_request() : nullptr;
  std::string verb = hx != nullptr && !hx->method.empty()
                         ? hx->method
                         : (body.empty() ? "GET" : "POST");
  std::string head;
  head.reserve(256);
  head += verb + " " + path + " HTTP/1.1\r\n";
  head += "Host: " + endpoint2str(cntl->remote_side()) + "\r\n";
  head += "User-Agent: brpc-amd/1.0\r\n";
  head += "Accept: */*\r\n";
  if (hx != nullptr) {
    for (const auto& kv : hx->headers) head += kv.first + ": " + kv.second + "\r\n";
  }
  if (!body.empty() || (hx != nullptr && !hx->content_type.empty())) {
    head += "Content-Type: " +
            (hx != nullptr && !hx->content_type.empty() ? hx->content_type
                                                        : "application/octet-stream") +
            "\r\n";
    head += "Content-Length: " + std::to_string(body.size()) + "\r\n";
  }
  head += "\r\n";
  out->append(head);
  out->append(body);
}

ParseResult ParseHttpMessage(IOBuf* source, Socket* sock, bool eof) {
  // Client-side sockets (no owning server) carry responses.
  if (sock->user() == nullptr) return ParseHttpResponse(source, sock, eof);
  // Cheap probe for an HTTP method prefix.
  char probe[8];
  size_t n = std::min<size_t>(source->size(), 8);
  if (n < 4) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* p = (const char*)source->fetch(probe, n);
  static const char* kMethods[] = {"GET ", "POST", "PUT ", "DELE", "HEAD", "OPTI", "PATC"};
  bool maybe = false;
  for (const char* m : kMethods) {
    if (memcmp(p, m, 4) == 0) {
      maybe = true;
      break;
    }
  }
  if (!maybe) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  // Copy up to 64 KiB looking for header end.
  std::string head;
  size_t scan = std::min<size_t>(source->size(), 64 << 10);
  source->copy_to(&head, scan, 0);
  size_t hend = find_header_end(head);
  if (hend == std::string::npos) {
    if (scan >= (64 << 10)) return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  }
  HttpMessage* msg = new HttpMessage;
  if (!parse_headers(head.substr(0, hend), &msg->req)) {
    delete msg;
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  }
  size_t content_len = 0;
  auto it = msg->req.headers.find("content-length");
  if (it != msg->req.headers.end() && !parse_body_size(it->second.c_str(), &content_len)) {
    delete msg;
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  }
  if (source->size() < hend + content_len) {
    delete msg;
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  }
  source->pop_front(hend);
  source->cutn(&msg->req.body, content_len);
  auto conn = msg->req.headers.find("connection");
  msg->req.keep_alive = !(conn != msg->req.headers.end() && conn->second == "close");
  return ParseResult::make_ok(msg);
}

const char* status_reason(int code) {
  switch (code) {
    case 200: return "OK";
    case 400: return "Bad Request";
    case 403: return "Forbidden";
    case 404: return "Not Found";
    case 500: return "Internal Server Error";
    case 503: return "Service Unavailable";
    default: return "Unknown";
  }
}

// Serializes and writes one HTTP/1.1 response. Addressed by SocketId so it
// is safe to run from ANY thread at ANY time (async handlers complete after
// ProcessHttpRequest returned; the socket may have died meanwhile).
void SendHttpResponseToSocket(SocketId sid, HttpResponse* resp, bool keep_alive) {
  SocketUniquePtr sock;
  if (Socket::Address(sid, &sock) != 0) return;
  std::string head;
  head.reserve(256);
  head += "HTTP/1.1 " + std::to_string(resp->status) + " " + status_reason(resp->status) +
          "\r\n";
  head += "Content-Type: " + resp->content_type + "\r\n";
  head += "Content-Length: " + std::to_string(resp->body.size()) + "\r\n";
  for (const auto& kv : resp->headers) head += kv.first + ": " + kv.second + "\r\n";
  if (!keep_alive) head += "Connection: close\r\n";
  head += "\r\n";
  IOBuf out;
  out.append(head);
  out.append(std::move(resp->body));
  sock->Write(&out);
  if (!keep_alive) sock->SetFailed(0, "connection: close");
}

void ProcessHttpRequest(InputMessageBase* msg_base) {
  HttpMessage* msg = (HttpMessage*)msg_base;
  SocketUniquePtr sock;
  if (Socket::Address(msg->socket_id, &sock) != 0) {
    delete msg;
    return;
  }
  Server* server = (Server*)sock->user();
  HttpResponse resp;
  resp.status = 200;
  resp.content_type = "text/plain";

  bool handled = DispatchBuiltinService(server, msg->req, &resp);
  if (!handled && server != nullptr) {
    // restful mappings first (AddService(..., "/v1/x => Method")), then
    // the default /Service/Method form
    std::string svc, method;
    if (!server->MapRestfulPath(msg->req.path, &svc, &method)) {
      std::string path = msg->req.path;
      if (!path.empty() && path[0] == '/') path = path.substr(1);
      size_t slash = path.find('/');
      svc = slash == std::string::npos ? "" : path.substr(0, slash);
      method = slash == std::string::npos ? path : path.substr(slash + 1);
    }
    const MethodFn* fn = server->FindMethod(svc, method);
    if (fn != nullptr) {
      // Heap-allocated call state: async handlers may run `done` from
      // another fiber long after this function returned (the reference
      // sends the response from the done closure too; a stack capture
      // here was a use-after-free).
      struct HttpCallCtx {
        Controller cntl;
        IOBuf resp_body;
        SocketId sid;
        bool keep_alive;
      };
      HttpCallCtx* ctx = new HttpCallCtx;
      ctx->cntl.server_ = server;
      ctx->cntl.server_socket_ = sock->id();
      ctx->cntl.remote_side_ = sock->remote_side();
      HttpHeaderExt& hreq = ctx->cntl.http_request();
      hreq.method = msg->req.method;
      for (const auto& kv : msg->req.headers) hreq.headers[kv.first] = kv.second;
      auto ct = msg->req.headers.find("content-type");
      if (ct != msg->req.headers.end()) hreq.content_type = ct->second;
      ctx->sid = sock->id();
      ctx->keep_alive = msg->req.keep_alive;
      Closure* done = NewCallback([ctx] {
        HttpResponse r;
        r.status = 200;
        if (ctx->cntl.Failed()) {
          r.status = 500;
          r.content_type = "text/plain";
          r.body.append(ctx->cntl.ErrorText());
        } else {
          r.content_type = "application/octet-stream";
          // Handler-set response view (status/content-type/headers).
          if (ctx->cntl.has_http_response()) {
            const HttpHeaderExt& hx = ctx->cntl.http_response();
            if (hx.status_code != 0) r.status = hx.status_code;
            if (!hx.content_type.empty()) r.content_type = hx.content_type;
            for (const auto& kv : hx.headers) r.headers[kv.first] = kv.second;
          }
          r.body.append(std::move(ctx->resp_body));
        }
        SendHttpResponseToSocket(ctx->sid, &r, ctx->keep_alive);
        delete ctx;
      });
      (*fn)(&ctx->cntl, msg->req.body, &ctx->resp_body, done);
      delete msg;
      return;
    }
  }
  if (!handled) {
    resp.status = 404;
    resp.body.append("no such page/method: " + msg->req.path + "\n");
  }
  SendHttpResponseToSocket(msg->socket_id, &resp, msg->req.keep_alive);
  delete msg;
}

}  // namespace

void RegisterHttpProtocol() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    Protocol p;
    p.parse = ParseHttpMessage;
    p.process_request = ProcessHttpRequest;
    p.process_response = ProcessHttpResponse;
    p.pack_request = PackHttp1Request;
    p.client_pipelined = true;  // HTTP/1.1 responses match requests FIFO
    p.support_server = true;
    p.support_client = true;
    p.name = "http";
    g_http_protocol_index = RegisterProtocol(p);
  });
}

bool ParseHttpHead(const std::string& head, HttpRequest* out) {
  return parse_headers(head, out);
}

}  // namespace policy
}  // namespace bam
