// brpc_amd: legacy Baidu wire protocols — hulu_pbrpc, sofa_pbrpc, nshead.
// Parity (clean-room from observed formats, no code copied):
//  - hulu_pbrpc (reference policy/hulu_pbrpc_protocol.cpp:50): 12-byte
//    header [HULU][body_size u32 LE][meta_size u32 LE]; body = meta
//    (HuluRpcRequestMeta/HuluRpcResponseMeta, policy/hulu_pbrpc_meta.proto)
//    + payload. NOTE: little-endian on purpose — hulu never used network
//    byte order. Response correlation_id is sint64 (zigzag).
//  - sofa_pbrpc (reference policy/sofa_pbrpc_protocol.cpp:49): 24-byte
//    header [SOFA][meta_size u32][body_size u64][msg_size u64], all LE,
//    msg_size == meta_size + body_size; meta = SofaRpcMeta
//    (policy/sofa_pbrpc_meta.proto: type=1, sequence_id=2, method=100,
//    failed=200, error_code=201, reason=202, compress_type=300).
//  - nshead (reference nshead.h:28): 36-byte little-endian struct
//    {id u16, version u16, log_id u32, provider[16], magic u32 0xfb709394,
//    reserved u32, body_len u32} + raw body. No correlation id: responses
//    match requests FIFO per connection (client_pipelined). The server
//    passes the raw body to ServerOptions::nshead_handler.
// Deltas: hulu credential_data / chunk info are not interpreted; method
// dispatch uses method_name (field 14), which our client always sends.
#include <string.h>

#include <mutex>
#include <string>

#include "base/iobuf.h"
#include "base/mcpack.h"
#include "base/logging.h"
#include "base/time.h"
#include "rpc/compress.h"
#include "rpc/controller.h"
#include "rpc/protocol.h"
#include "rpc/rpc_errno.h"
#include "rpc/server.h"
#include "rpc/socket.h"
#include "rpc/wire.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

namespace policy {

namespace {

inline void put_u32_le(std::string* out, uint32_t v) {
  char b[4];
  memcpy(b, &v, 4);  // x86-64: host order IS little-endian
  out->append(b, 4);
}
inline void put_u64_le(std::string* out, uint64_t v) {
  char b[8];
  memcpy(b, &v, 8);
  out->append(b, 8);
}
inline uint32_t get_u32_le(const char* p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;
}
inline uint64_t get_u64_le(const char* p) {
  uint64_t v;
  memcpy(&v, p, 8);
  return v;
}

inline uint64_t zigzag_enc(int64_t v) { return ((uint64_t)v << 1) ^ (uint64_t)(v >> 63); }
inline int64_t zigzag_dec(uint64_t v) { return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); }

// Generic request-execution helper shared by hulu/sofa servers: looks up
// the method, runs it, replies via `reply` (which owns cntl/resp).
struct LegacyCall {
  Controller* cntl;
  IOBuf* resp;
  Closure* done;
};

bool start_server_call(Server* server, SocketId sid, const std::string& service,
                       const std::string& method, IOBuf* req_payload, int compress_type,
                       LegacyCall* out, void (*send)(SocketId, Controller*, IOBuf*, void*),
                       void* send_arg) {
  Controller* cntl = new Controller;
  cntl->server_ = server;
  cntl->server_socket_ = sid;
  cntl->call.service_name = service;
  cntl->call.method_name = method;
  IOBuf* resp = new IOBuf;
  Closure* done =
      NewCallback([sid, cntl, resp, send, send_arg] { send(sid, cntl, resp, send_arg); });
  if (compress_type != 0) {
    RegisterBuiltinCompressHandlers();
    IOBuf plain;
    if (!ApplyDecompress((CompressType)compress_type, *req_payload, &plain)) {
      cntl->SetFailed(EREQUEST, "request decompression failed");
      done->Run();
      return false;
    }
    req_payload->swap(plain);
  }
  Service* svc = nullptr;
  const MethodFn* fn =
      server != nullptr ? server->FindMethod(service, method, &svc) : nullptr;
  if (fn == nullptr) {
    cntl->SetFailed(ENOMETHOD, "unknown method " + service + "." + method);
    done->Run();
    return false;
  }
  server->concurrency.fetch_add(1, std::memory_order_relaxed);
  cntl->concurrency_counted_ = true;
  (*fn)(cntl, *req_payload, resp, done);
  out->cntl = cntl;
  out->resp = resp;
  out->done = done;
  return true;
}

void finish_server_send(Controller* cntl, IOBuf* resp, IOBuf* packet, SocketId sid) {
  SocketUniquePtr sock;
  if (Socket::Address(sid, &sock) == 0) sock->Write(packet);
  if (cntl->server_ != nullptr) {
    if (cntl->concurrency_counted_)
      cntl->server_->concurrency.fetch_sub(1, std::memory_order_relaxed);
    cntl->server_->nprocessed.fetch_add(1, std::memory_order_relaxed);
  }
  delete resp;
  delete cntl;
}

// Completes a client call from a parsed (error_code, error_text, payload).
void finish_client_call(SessionId cid, SocketId socket_id, int error_code,
                        const std::string& error_text, IOBuf* payload, int compress_type) {
  void* data = nullptr;
  if (session_lock(cid, &data) != 0) return;  // late/duplicate
  Controller* cntl = (Controller*)data;
  {
    SocketUniquePtr sock;
    if (Socket::Address(socket_id, &sock) == 0) sock->remove_pending_session(cid);
  }
  if (error_code != 0) {
    cntl->SetFailed(error_code, error_text);
  } else if (cntl->call.response != nullptr) {
    cntl->call.response->clear();
    cntl->call.response->swap(*payload);
    if (compress_type != 0) {
      RegisterBuiltinCompressHandlers();
      IOBuf plain;
      if (ApplyDecompress((CompressType)compress_type, *cntl->call.response, &plain)) {
        cntl->call.response->swap(plain);
      } else {
        cntl->SetFailed(ERESPONSE, "response decompression failed");
      }
    }
  }
  EndRPC(cntl, cid);
}

// ==================== hulu_pbrpc ====================

constexpr char kHuluMagic[4] = {'H', 'U', 'L', 'U'};

struct HuluMessage : public InputMessageBase {
  // request meta
  std::string service_name;
  std::string method_name;
  int32_t method_index = -1;
  // response meta
  int error_code = 0;
  std::string error_text;
  bool is_response = false;
  int64_t correlation_id = 0;
  int compress_type = 0;
  uint64_t log_id = 0;
  IOBuf payload;
};

ParseResult ParseHulu(IOBuf* source, Socket*, bool) {
  char aux[12];
  if (source->size() < 12) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(aux, 12);
  if (h == nullptr || memcmp(h, kHuluMagic, 4) != 0)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  uint32_t body_size = get_u32_le(h + 4);
  uint32_t meta_size = get_u32_le(h + 8);
  if (meta_size > body_size || body_size > (256u << 20))
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  if (source->size() < 12 + (size_t)body_size)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  source->pop_front(12);
  std::string meta_bytes;
  source->cutn(&meta_bytes, meta_size);
  HuluMessage* msg = new HuluMessage;
  wire::Reader r(meta_bytes.data(), meta_bytes.size());
  int wt;
  // The same header frames both directions; we decode both field sets and
  // decide request-vs-response from which of them appeared (the server
  // never sends service_name; the client never sends error_code). A
  // response is identified by parse context anyway (socket side).
  bool saw_request_fields = false;
  for (int f; (f = r.read_tag(&wt)) != 0;) {
    switch (f) {
      case 1:
        // request: service_name (string) | response: error_code (varint)
        if (wt == 2) {
          msg->service_name = r.read_string();
          saw_request_fields = true;
        } else {
          msg->error_code = (int)r.varint();
          msg->is_response = true;
        }
        break;
      case 2:
        if (wt == 0) {
          msg->method_index = (int32_t)r.varint();
          saw_request_fields = true;
        } else {
          msg->error_text = r.read_string();
          msg->is_response = true;
        }
        break;
      case 3:
        if (msg->is_response || !saw_request_fields) {
          msg->correlation_id = zigzag_dec(r.varint());  // response sint64
          msg->is_response = true;
        } else {
          msg->compress_type = (int)r.varint();  // request
        }
        break;
      case 4:
        if (msg->is_response) msg->compress_type = (int)r.varint();
        else msg->correlation_id = (int64_t)r.varint();
        break;
      case 5:
        msg->log_id = r.varint();
        break;
      case 14:
        msg->method_name = r.read_string();
        break;
      default:
        r.skip(wt);
    }
    if (!r.ok()) {
      delete msg;
      return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
    }
  }
  source->cutn(&msg->payload, body_size - meta_size);
  return ParseResult::make_ok(msg);
}

// hulu compress codes: 1 = snappy, 2 = gzip (ours: SNAPPY=1, GZIP=2 — same).

void SendHuluResponse(SocketId sid, Controller* cntl, IOBuf* resp, void* arg) {
  int64_t correlation_id = (int64_t)(uintptr_t)arg;
  std::string meta;
  if (cntl->ErrorCode() != 0) wire::put_int_field(&meta, 1, cntl->ErrorCode());
  if (!cntl->ErrorText().empty()) wire::put_str_field(&meta, 2, cntl->ErrorText());
  // field 3: sint64 correlation id (zigzag)
  wire::put_int_field(&meta, 3, (int64_t)zigzag_enc(correlation_id));
  IOBuf packet;
  std::string header;
  header.append(kHuluMagic, 4);
  put_u32_le(&header, (uint32_t)(meta.size() + resp->size()));
  put_u32_le(&header, (uint32_t)meta.size());
  packet.append(header);
  packet.append(meta);
  packet.append(*resp);
  finish_server_send(cntl, resp, &packet, sid);
}

void ProcessHuluRequest(InputMessageBase* mb) {
  HuluMessage* msg = (HuluMessage*)mb;
  SocketUniquePtr sock;
  if (Socket::Address(msg->socket_id, &sock) != 0) {
    delete msg;
    return;
  }
  Server* server = (Server*)sock->user();
  LegacyCall call;
  start_server_call(server, msg->socket_id, msg->service_name, msg->method_name,
                    &msg->payload, msg->compress_type, &call, SendHuluResponse,
                    (void*)(uintptr_t)msg->correlation_id);
  delete msg;
}

void ProcessHuluResponse(InputMessageBase* mb) {
  HuluMessage* msg = (HuluMessage*)mb;
  finish_client_call((SessionId)msg->correlation_id, msg->socket_id, msg->error_code,
                     msg->error_text, &msg->payload, msg->compress_type);
  delete msg;
}

void PackHuluRequest(IOBuf* out, Controller* cntl, uint64_t correlation_id) {
  std::string meta;
  wire::put_str_field(&meta, 1, cntl->call.service_name);
  wire::put_int_field(&meta, 2, 0);  // method_index (we dispatch by name)
  if (cntl->request_compress_type() != COMPRESS_TYPE_NONE)
    wire::put_int_field(&meta, 3, (int)cntl->request_compress_type());
  wire::put_int_field(&meta, 4, (int64_t)correlation_id);
  if (cntl->log_id() != 0) wire::put_int_field(&meta, 5, (int64_t)cntl->log_id());
  wire::put_str_field(&meta, 14, cntl->call.method_name);
  IOBuf body = cntl->call.request_buf;
  if (cntl->request_compress_type() != COMPRESS_TYPE_NONE) {
    RegisterBuiltinCompressHandlers();
    IOBuf compressed;
    if (ApplyCompress(cntl->request_compress_type(), body, &compressed)) body.swap(compressed);
  }
  std::string header;
  header.append(kHuluMagic, 4);
  put_u32_le(&header, (uint32_t)(meta.size() + body.size()));
  put_u32_le(&header, (uint32_t)meta.size());
  out->append(header);
  out->append(meta);
  out->append(std::move(body));
}

// ==================== sofa_pbrpc ====================

constexpr char kSofaMagic[4] = {'S', 'O', 'F', 'A'};

struct SofaMessage : public InputMessageBase {
  int type = 0;  // 0 request, 1 response
  uint64_t sequence_id = 0;
  std::string method;  // "pkg.Service.Method"
  bool failed = false;
  int error_code = 0;
  std::string reason;
  int compress_type = 0;  // sofa codes
  IOBuf payload;
};

ParseResult ParseSofa(IOBuf* source, Socket*, bool) {
  char aux[24];
  if (source->size() < 24) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(aux, 24);
  if (h == nullptr || memcmp(h, kSofaMagic, 4) != 0)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  uint32_t meta_size = get_u32_le(h + 4);
  uint64_t body_size = get_u64_le(h + 8);
  uint64_t msg_size = get_u64_le(h + 16);
  if (msg_size != meta_size + body_size || msg_size > (256u << 20))
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  if (source->size() < 24 + msg_size)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  source->pop_front(24);
  std::string meta_bytes;
  source->cutn(&meta_bytes, meta_size);
  SofaMessage* msg = new SofaMessage;
  wire::Reader r(meta_bytes.data(), meta_bytes.size());
  int wt;
  for (int f; (f = r.read_tag(&wt)) != 0;) {
    switch (f) {
      case 1:
        msg->type = (int)r.varint();
        break;
      case 2:
        msg->sequence_id = r.varint();
        break;
      case 100:
        msg->method = r.read_string();
        break;
      case 200:
        msg->failed = r.varint() != 0;
        break;
      case 201:
        msg->error_code = (int)r.varint();
        break;
      case 202:
        msg->reason = r.read_string();
        break;
      case 300:
        msg->compress_type = (int)r.varint();
        break;
      default:
        r.skip(wt);
    }
    if (!r.ok()) {
      delete msg;
      return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
    }
  }
  source->cutn(&msg->payload, body_size);
  return ParseResult::make_ok(msg);
}

// sofa compress codes: 1 gzip, 2 zlib, 3 snappy, 4 lz4.
int to_sofa_compress(CompressType t) {
  switch (t) {
    case COMPRESS_TYPE_SNAPPY: return 3;
    case COMPRESS_TYPE_GZIP: return 1;
    default: return 0;
  }
}
CompressType from_sofa_compress(int c) {
  switch (c) {
    case 3: return COMPRESS_TYPE_SNAPPY;
    case 1: return COMPRESS_TYPE_GZIP;
    default: return COMPRESS_TYPE_NONE;
  }
}

void pack_sofa_frame(IOBuf* out, const std::string& meta, IOBuf&& body) {
  std::string header;
  header.append(kSofaMagic, 4);
  put_u32_le(&header, (uint32_t)meta.size());
  put_u64_le(&header, (uint64_t)body.size());
  put_u64_le(&header, (uint64_t)(meta.size() + body.size()));
  out->append(header);
  out->append(meta);
  out->append(std::move(body));
}

void SendSofaResponse(SocketId sid, Controller* cntl, IOBuf* resp, void* arg) {
  uint64_t seq = (uint64_t)(uintptr_t)arg;
  std::string meta;
  wire::put_int_field(&meta, 1, 1);  // type = RESPONSE
  wire::put_int_field(&meta, 2, (int64_t)seq);
  if (cntl->ErrorCode() != 0) {
    wire::put_int_field(&meta, 200, 1);  // failed
    wire::put_int_field(&meta, 201, cntl->ErrorCode());
    wire::put_str_field(&meta, 202, cntl->ErrorText());
  }
  IOBuf packet;
  pack_sofa_frame(&packet, meta, std::move(*resp));
  finish_server_send(cntl, resp, &packet, sid);
}

void ProcessSofaRequest(InputMessageBase* mb) {
  SofaMessage* msg = (SofaMessage*)mb;
  SocketUniquePtr sock;
  if (Socket::Address(msg->socket_id, &sock) != 0) {
    delete msg;
    return;
  }
  Server* server = (Server*)sock->user();
  // "pkg.Service.Method" -> service = all-but-last, method = last
  std::string service = msg->method, method;
  size_t dot = msg->method.find_last_of('.');
  if (dot != std::string::npos) {
    service = msg->method.substr(0, dot);
    method = msg->method.substr(dot + 1);
  } else {
    method = msg->method;
    service.clear();
  }
  LegacyCall call;
  start_server_call(server, msg->socket_id, service, method, &msg->payload,
                    (int)from_sofa_compress(msg->compress_type), &call, SendSofaResponse,
                    (void*)(uintptr_t)msg->sequence_id);
  delete msg;
}

void ProcessSofaResponse(InputMessageBase* mb) {
  SofaMessage* msg = (SofaMessage*)mb;
  finish_client_call((SessionId)msg->sequence_id, msg->socket_id,
                     msg->failed ? (msg->error_code != 0 ? msg->error_code : ERESPONSE) : 0,
                     msg->reason, &msg->payload,
                     (int)from_sofa_compress(msg->compress_type));
  delete msg;
}

void PackSofaRequest(IOBuf* out, Controller* cntl, uint64_t correlation_id) {
  std::string meta;
  wire::put_int_field(&meta, 1, 0);  // type = REQUEST
  wire::put_int_field(&meta, 2, (int64_t)correlation_id);
  std::string full = cntl->call.service_name.empty()
                         ? cntl->call.method_name
                         : cntl->call.service_name + "." + cntl->call.method_name;
  wire::put_str_field(&meta, 100, full);
  IOBuf body = cntl->call.request_buf;
  if (cntl->request_compress_type() != COMPRESS_TYPE_NONE) {
    RegisterBuiltinCompressHandlers();
    IOBuf compressed;
    if (ApplyCompress(cntl->request_compress_type(), body, &compressed)) {
      body.swap(compressed);
      wire::put_int_field(&meta, 300, to_sofa_compress(cntl->request_compress_type()));
    }
  }
  pack_sofa_frame(out, meta, std::move(body));
}

// ==================== nshead ====================

constexpr uint32_t kNsheadMagic = 0xfb709394u;
constexpr size_t kNsheadLen = 36;

int g_nshead_protocol_index = -1;
int g_nova_protocol_index = -1;

struct NsheadMessage : public InputMessageBase {
  uint16_t id = 0;
  uint16_t version = 0;
  uint32_t log_id = 0;
  char provider[16] = {0};
  IOBuf body;
};

// Frame-only parse shared by nshead / nova / ubrpc (no ownership gate).
ParseResult parse_nshead_frame(IOBuf* source) {
  char aux[kNsheadLen];
  if (source->size() < kNsheadLen) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(aux, kNsheadLen);
  if (h == nullptr || get_u32_le(h + 24) != kNsheadMagic)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  uint32_t body_len = get_u32_le(h + 32);
  if (body_len > (256u << 20)) return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  if (source->size() < kNsheadLen + body_len)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  NsheadMessage* msg = new NsheadMessage;
  memcpy(&msg->id, h, 2);
  memcpy(&msg->version, h + 2, 2);
  memcpy(&msg->log_id, h + 4, 4);
  memcpy(msg->provider, h + 8, 16);
  source->pop_front(kNsheadLen);
  source->cutn(&msg->body, body_len);
  return ParseResult::make_ok(msg);
}

ParseResult ParseNshead(IOBuf* source, Socket* sock, bool) {
  // Client side: ubrpc shares this frame with its OWN id-correlated parse —
  // the FIFO path only claims connections created for nshead/nova.
  if (sock->user() == nullptr && sock->client_protocol_hint != g_nshead_protocol_index &&
      sock->client_protocol_hint != g_nova_protocol_index) {
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  }
  return parse_nshead_frame(source);
}

void pack_nshead(IOBuf* out, uint16_t id, uint16_t version, uint32_t log_id,
                 const char provider[16], const IOBuf& body) {
  char h[kNsheadLen];
  memset(h, 0, sizeof(h));
  memcpy(h, &id, 2);
  memcpy(h + 2, &version, 2);
  memcpy(h + 4, &log_id, 4);
  if (provider != nullptr) memcpy(h + 8, provider, 16);
  uint32_t magic = kNsheadMagic;
  memcpy(h + 24, &magic, 4);
  uint32_t blen = (uint32_t)body.size();
  memcpy(h + 32, &blen, 4);
  out->append(h, kNsheadLen);
  out->append(body);
}

void ProcessNsheadRequest(InputMessageBase* mb) {
  NsheadMessage* msg = (NsheadMessage*)mb;
  SocketUniquePtr sock;
  if (Socket::Address(msg->socket_id, &sock) != 0) {
    delete msg;
    return;
  }
  Server* server = (Server*)sock->user();
  IOBuf resp_body;
  if (server != nullptr && server->options().nshead_handler) {
    server->options().nshead_handler(msg->body, &resp_body);
    server->nprocessed.fetch_add(1, std::memory_order_relaxed);
  }  // no handler: empty response body (the connection stays usable)
  IOBuf packet;
  pack_nshead(&packet, msg->id, msg->version, msg->log_id, msg->provider, resp_body);
  sock->Write(&packet);
  delete msg;
}

void ProcessNsheadResponse(InputMessageBase* mb) {
  NsheadMessage* msg = (NsheadMessage*)mb;
  SocketUniquePtr sock;
  SessionId cid = 0;
  if (Socket::Address(msg->socket_id, &sock) == 0) cid = sock->pop_pipeline();
  if (cid != 0) finish_client_call(cid, msg->socket_id, 0, "", &msg->body, 0);
  delete msg;
}

void PackNsheadRequest(IOBuf* out, Controller* cntl, uint64_t /*fifo-correlated*/) {
  pack_nshead(out, 0, 0, (uint32_t)cntl->log_id(), nullptr, cntl->call.request_buf);
}

// ==================== nova_pbrpc (client) ====================
// Parity: reference policy/nova_pbrpc_protocol.cpp — nshead framing with
// NO meta message: nshead.reserved carries the method index,
// nshead.version==1 marks a snappy-compressed protobuf body; responses
// come back nshead-framed, FIFO-correlated (nshead has no correlation
// id). Server side parity is the NsheadService adaptor
// (ServerOptions::nshead_handler), exactly like the reference's
// NovaServiceAdaptor over NsheadService.

void PackNovaRequest(IOBuf* out, Controller* cntl, uint64_t /*fifo*/) {
  uint32_t method_index = (uint32_t)strtoul(cntl->call.method_name.c_str(), nullptr, 10);
  IOBuf body = cntl->call.request_buf;
  uint16_t version = 0;
  if (cntl->request_compress_type() == COMPRESS_TYPE_SNAPPY) {
    RegisterBuiltinCompressHandlers();
    IOBuf compressed;
    if (ApplyCompress(COMPRESS_TYPE_SNAPPY, body, &compressed)) {
      body.swap(compressed);
      version = 1;
    }
  }
  char h[36];
  memset(h, 0, sizeof(h));
  memcpy(h + 2, &version, 2);
  uint32_t log_id = (uint32_t)cntl->log_id();
  memcpy(h + 4, &log_id, 4);
  uint32_t magic = 0xfb709394u;
  memcpy(h + 24, &magic, 4);
  memcpy(h + 28, &method_index, 4);  // reserved = method index
  uint32_t blen = (uint32_t)body.size();
  memcpy(h + 32, &blen, 4);
  out->append(h, sizeof(h));
  out->append(std::move(body));
}

// ==================== public_pbrpc (client) ====================
// Parity: reference policy/public_pbrpc_protocol.cpp +
// public_pbrpc_meta.proto — nshead (version 1000, provider "__pbrpc__")
// whose body is ONE protobuf `PublicPbrpcRequest`
// {requestHead{...}=1, requestBody{version=1 "pbrpc=1.0", service=3,
// method_id=4, id=5, serialized_request=6}=2}; responses are
// `PublicPbrpcResponse` {responseHead{code=1 sint32, text=2}=1,
// responseBody{serialized_response=1, error=3, id=4}=2}. Correlated by
// body.id (hand-rolled codec via rpc/wire.h, like baidu_std's meta).

int g_public_protocol_index = -1;

struct PublicMessage : public InputMessageBase {
  int code = 0;          // head.code or body.error
  std::string text;
  uint64_t id = 0;
  IOBuf payload;         // serialized_response
};

ParseResult ParsePublic(IOBuf* source, Socket* sock, bool) {
  if (sock->client_protocol_hint != g_public_protocol_index)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  ParseResult raw = parse_nshead_frame(source);
  if (raw.error != PARSE_OK) return raw;
  NsheadMessage* nm = (NsheadMessage*)raw.msg;
  std::string body = nm->body.to_string();
  delete nm;
  PublicMessage* msg = new PublicMessage;
  wire::Reader r(body.data(), body.size());
  int wt;
  bool ok = true;
  for (int f; ok && (f = r.read_tag(&wt)) != 0;) {
    if (f == 1) {  // responseHead
      std::string sub = r.read_string();
      wire::Reader rh(sub.data(), sub.size());
      int wt2;
      for (int f2; (f2 = rh.read_tag(&wt2)) != 0;) {
        if (f2 == 1) msg->code = (int)zigzag_dec(rh.varint());  // sint32
        else if (f2 == 2) msg->text = rh.read_string();
        else rh.skip(wt2);
        if (!rh.ok()) { ok = false; break; }
      }
    } else if (f == 2) {  // responseBody
      std::string sub = r.read_string();
      wire::Reader rb(sub.data(), sub.size());
      int wt2;
      for (int f2; (f2 = rb.read_tag(&wt2)) != 0;) {
        if (f2 == 1) {
          std::string sr = rb.read_string();
          msg->payload.clear();
          msg->payload.append(sr);
        } else if (f2 == 3) {
          int err = (int)rb.varint();
          if (err != 0) msg->code = err;
        } else if (f2 == 4) {
          msg->id = rb.varint();
        } else {
          rb.skip(wt2);
        }
        if (!rb.ok()) { ok = false; break; }
      }
    } else {
      r.skip(wt);
    }
    if (!r.ok()) ok = false;
  }
  if (!ok) {
    delete msg;
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  }
  return ParseResult::make_ok(msg);
}

void ProcessPublicResponse(InputMessageBase* mb) {
  PublicMessage* msg = (PublicMessage*)mb;
  finish_client_call((SessionId)msg->id, msg->socket_id, msg->code, msg->text, &msg->payload,
                     0);
  delete msg;
}

void PackPublicRequest(IOBuf* out, Controller* cntl, uint64_t correlation_id) {
  std::string head;
  wire::put_str_field(&head, 1, "127.0.0.1");        // from_host
  wire::put_int_field(&head, 3, 1);                   // connection = true
  if (cntl->log_id() != 0) wire::put_int_field(&head, 7, (int64_t)cntl->log_id());
  std::string bodymsg;
  wire::put_str_field(&bodymsg, 1, "pbrpc=1.0");      // version
  wire::put_str_field(&bodymsg, 3, cntl->call.service_name);
  wire::put_int_field(&bodymsg, 4,
                      (int64_t)strtoul(cntl->call.method_name.c_str(), nullptr, 10));
  wire::put_int_field(&bodymsg, 5, (int64_t)correlation_id);
  wire::put_str_field(&bodymsg, 6, cntl->call.request_buf.to_string());
  std::string req;
  wire::put_msg_field(&req, 1, head);
  wire::put_msg_field(&req, 2, bodymsg);
  char h[36];
  memset(h, 0, sizeof(h));
  uint16_t version = 1000;
  memcpy(h + 2, &version, 2);
  uint32_t log_id32 = (uint32_t)cntl->log_id();
  memcpy(h + 4, &log_id32, 4);
  memcpy(h + 8, "__pbrpc__", 9);
  uint32_t magic = 0xfb709394u;
  memcpy(h + 24, &magic, 4);
  uint32_t blen = (uint32_t)req.size();
  memcpy(h + 32, &blen, 4);
  out->append(h, sizeof(h));
  out->append(req);
}

// ==================== ubrpc (client) ====================
// Parity: reference policy/ubrpc2pb_protocol.cpp — nshead (version 1000)
// framing a compack/mcpack object:
//   request:  {header:{...}, content:[{service_name, id, method, params{...}}]}
//   response: {content:[{id, (code,message) | result/params...}]}
// Correlation rides content[0].id (a REAL correlation id, unlike plain
// nshead's FIFO). We are payload-centric: params = {"req": <binary>}; the
// response payload handed back is the re-serialized content[0] object.

int g_ubrpc_protocol_index = -1;

struct UbrpcMessage : public InputMessageBase {
  int64_t id = 0;
  int code = 0;
  std::string message;
  IOBuf payload;  // mcpack of content[0]
};

ParseResult ParseUbrpc(IOBuf* source, Socket* sock, bool eof) {
  // reuse the nshead frame; gate on our own hint so plain nshead keeps
  // working independently
  if (sock->client_protocol_hint != g_ubrpc_protocol_index)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  (void)eof;
  ParseResult raw = parse_nshead_frame(source);
  if (raw.error != PARSE_OK) return raw;
  NsheadMessage* nm = (NsheadMessage*)raw.msg;
  std::string body = nm->body.to_string();
  delete nm;
  mcpack::Value root;
  if (!mcpack::Parse(body.data(), body.size(), &root) ||
      root.obj.count("content") == 0 || root.obj["content"].arr.empty()) {
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  }
  mcpack::Value& c0 = root.obj["content"].arr[0];
  UbrpcMessage* msg = new UbrpcMessage;
  auto it = c0.obj.find("id");
  if (it != c0.obj.end()) msg->id = it->second.type == mcpack::Value::UINT
                                        ? (int64_t)it->second.u
                                        : it->second.i;
  it = c0.obj.find("code");
  if (it != c0.obj.end()) msg->code = (int)(it->second.type == mcpack::Value::UINT
                                                ? (int64_t)it->second.u
                                                : it->second.i);
  it = c0.obj.find("message");
  if (it != c0.obj.end()) msg->message = it->second.str;
  std::string re;
  mcpack::Serialize(c0.type == mcpack::Value::OBJECT ? c0 : mcpack::Value::Object(), &re);
  msg->payload.append(re);
  return ParseResult::make_ok(msg);
}

void ProcessUbrpcResponse(InputMessageBase* mb) {
  UbrpcMessage* msg = (UbrpcMessage*)mb;
  finish_client_call((SessionId)msg->id, msg->socket_id,
                     msg->code != 0 ? msg->code : 0, msg->message, &msg->payload, 0);
  delete msg;
}

void PackUbrpcRequest(IOBuf* out, Controller* cntl, uint64_t correlation_id) {
  mcpack::Value root = mcpack::Value::Object();
  mcpack::Value header = mcpack::Value::Object();
  header.obj["connection"] = mcpack::Value::Bool(true);
  root.obj["header"] = std::move(header);
  mcpack::Value c0 = mcpack::Value::Object();
  c0.obj["service_name"] = mcpack::Value::Str(cntl->call.service_name);
  c0.obj["method"] = mcpack::Value::Str(cntl->call.method_name);
  c0.obj["id"] = mcpack::Value::Int((int64_t)correlation_id);
  mcpack::Value params = mcpack::Value::Object();
  params.obj["req"] = mcpack::Value::Bin(cntl->call.request_buf.to_string());
  c0.obj["params"] = std::move(params);
  mcpack::Value content = mcpack::Value::Array();
  content.arr.push_back(std::move(c0));
  root.obj["content"] = std::move(content);
  std::string body;
  mcpack::Serialize(root, &body);
  char h[36];
  memset(h, 0, sizeof(h));
  uint16_t version = 1000;  // UBRPC_NSHEAD_VERSION
  memcpy(h + 2, &version, 2);
  uint32_t magic = 0xfb709394u;
  memcpy(h + 24, &magic, 4);
  uint32_t blen = (uint32_t)body.size();
  memcpy(h + 32, &blen, 4);
  out->append(h, sizeof(h));
  out->append(body);
}

// ==================== esp (client) ====================
// Parity: reference policy/esp_protocol.cpp + esp_head.h — 32-byte packed
// little-endian head {from u64, to u64, msg u32, msg_id u64, body_len i32}
// followed by a raw body. The head has NO magic, so the parse is gated to
// sockets created for the esp protocol (client_protocol_hint). Responses
// match requests FIFO per connection (the reference allows one pending
// call per pooled connection; FIFO is the superset of that).

#pragma pack(push, 1)
struct EspHead {
  uint64_t from = 0;
  uint64_t to = 0;
  uint32_t msg = 0;
  uint64_t msg_id = 0;
  int32_t body_len = 0;
};
#pragma pack(pop)
static_assert(sizeof(EspHead) == 32, "esp head is 32 packed bytes");

int g_esp_protocol_index = -1;

struct EspMsg : public InputMessageBase {
  EspHead head;
  IOBuf body;
};

ParseResult ParseEsp(IOBuf* source, Socket* sock, bool) {
  if (sock->client_protocol_hint != g_esp_protocol_index)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  char aux[sizeof(EspHead)];
  if (source->size() < sizeof(EspHead))
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(aux, sizeof(EspHead));
  EspHead head;
  memcpy(&head, h, sizeof(head));
  if (head.body_len < 0 || head.body_len > (64 << 20))
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  if (source->size() < sizeof(EspHead) + (size_t)head.body_len)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  EspMsg* msg = new EspMsg;
  msg->head = head;
  source->pop_front(sizeof(EspHead));
  source->cutn(&msg->body, (size_t)head.body_len);
  return ParseResult::make_ok(msg);
}

void ProcessEspResponse(InputMessageBase* mb) {
  EspMsg* msg = (EspMsg*)mb;
  SocketUniquePtr sock;
  SessionId cid = 0;
  if (Socket::Address(msg->socket_id, &sock) == 0) cid = sock->pop_pipeline();
  if (cid != 0) finish_client_call(cid, msg->socket_id, 0, "", &msg->body, 0);
  delete msg;
}

void PackEspRequest(IOBuf* out, Controller* cntl, uint64_t correlation_id) {
  EspHead head;
  // method name carries the numeric esp `msg` selector
  head.msg = (uint32_t)strtoul(cntl->call.method_name.c_str(), nullptr, 10);
  head.msg_id = correlation_id;
  head.body_len = (int32_t)cntl->call.request_buf.size();
  out->append(&head, sizeof(head));
  out->append(cntl->call.request_buf);
}

}  // namespace

void RegisterHuluProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParseHulu;
    p.process_request = ProcessHuluRequest;
    p.process_response = ProcessHuluResponse;
    p.pack_request = PackHuluRequest;
    p.support_server = true;
    p.support_client = true;
    p.name = "hulu_pbrpc";
    RegisterProtocol(p);
  });
}

void RegisterSofaProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParseSofa;
    p.process_request = ProcessSofaRequest;
    p.process_response = ProcessSofaResponse;
    p.pack_request = PackSofaRequest;
    p.support_server = true;
    p.support_client = true;
    p.name = "sofa_pbrpc";
    RegisterProtocol(p);
  });
}

void RegisterPublicPbrpcProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParsePublic;
    p.process_response = ProcessPublicResponse;
    p.pack_request = PackPublicRequest;
    p.support_server = false;
    p.support_client = true;
    p.name = "public_pbrpc";
    g_public_protocol_index = RegisterProtocol(p);
  });
}

void RegisterUbrpcProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParseUbrpc;
    p.process_response = ProcessUbrpcResponse;
    p.pack_request = PackUbrpcRequest;
    p.support_server = false;
    p.support_client = true;
    p.name = "ubrpc";
    g_ubrpc_protocol_index = RegisterProtocol(p);
  });
}

void RegisterNovaProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParseNshead;        // same frame; adaptor semantics differ
    p.process_response = ProcessNsheadResponse;
    p.pack_request = PackNovaRequest;
    p.client_pipelined = true;
    p.support_server = false;  // server side = nshead_handler adaptor
    p.support_client = true;
    p.name = "nova_pbrpc";
    g_nova_protocol_index = RegisterProtocol(p);
  });
}

void RegisterEspProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParseEsp;
    p.process_response = ProcessEspResponse;
    p.pack_request = PackEspRequest;
    p.client_pipelined = true;
    p.support_server = false;
    p.support_client = true;
    p.name = "esp";
    g_esp_protocol_index = RegisterProtocol(p);
  });
}

void RegisterNsheadProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParseNshead;
    p.process_request = ProcessNsheadRequest;
    p.process_response = ProcessNsheadResponse;
    p.pack_request = PackNsheadRequest;
    p.client_pipelined = true;  // no correlation id in the header
    p.support_server = true;
    p.support_client = true;
    p.name = "nshead";
    g_nshead_protocol_index = RegisterProtocol(p);
  });
}

}  // namespace policy
}  // namespace bam
