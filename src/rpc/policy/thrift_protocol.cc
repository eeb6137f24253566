// brpc_amd: Thrift framed-transport protocol (TBinaryProtocol envelope).
// Parity: reference brpc/thrift_*.cpp + policy/thrift_protocol.cpp:
// framed transport (u32 length prefix) + TBinary message header
// (version|type, method name, seqid); the user struct payload passes
// through opaquely (handlers produce/consume raw TBinary struct bytes).
// Server: methods registered under service "thrift"; client: pipelined
// FIFO correlation like redis/memcache.
#include <mutex>

#include "base/logging.h"
#include "fiber/session.h"
#include "rpc/channel.h"
#include "rpc/controller.h"
#include "rpc/server.h"
#include "rpc/wire.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

namespace policy {

namespace {

const uint32_t kThriftVersion1 = 0x80010000;
enum ThriftMsgType { T_CALL = 1, T_REPLY = 2, T_EXCEPTION = 3, T_ONEWAY = 4 };

struct ThriftMessage : public InputMessageBase {
  std::string method;
  int32_t seqid = 0;
  int msg_type = 0;
  IOBuf payload;  // raw TBinary struct bytes (args or result)
};

void PackThriftMessage(IOBuf* out, int msg_type, const std::string& method, int32_t seqid,
                       const IOBuf& payload) {
  std::string head;
  char b4[4];
  wire::put_u32_be(b4, kThriftVersion1 | (uint32_t)msg_type);
  head.append(b4, 4);
  wire::put_u32_be(b4, (uint32_t)method.size());
  head.append(b4, 4);
  head.append(method);
  wire::put_u32_be(b4, (uint32_t)seqid);
  head.append(b4, 4);
  uint32_t frame_len = (uint32_t)(head.size() + payload.size());
  char lenb[4];
  wire::put_u32_be(lenb, frame_len);
  out->append(lenb, 4);
  out->append(head);
  out->append(payload);
}

ParseResult ParseThrift(IOBuf* source, Socket* sock, bool /*eof*/) {
  char aux[16];
  if (source->size() < 12) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(aux, 12);
  uint32_t frame_len = wire::get_u32_be(h);
  uint32_t version = wire::get_u32_be(h + 4);
  if ((version & 0xffff0000) != kThriftVersion1)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  if (frame_len > (64u << 20) || frame_len < 8)
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  if (source->size() < 4 + frame_len)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const bool server_side = sock->user() != nullptr;
  int msg_type = (int)(version & 0xff);
  if (server_side && msg_type != T_CALL && msg_type != T_ONEWAY)
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  source->pop_front(8);  // frame len + version
  std::string name_len_raw;
  source->cutn(&name_len_raw, 4);
  uint32_t name_len = wire::get_u32_be(name_len_raw.data());
  if (name_len > frame_len) return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  ThriftMessage* msg = new ThriftMessage;
  msg->msg_type = msg_type;
  source->cutn(&msg->method, name_len);
  std::string seq_raw;
  source->cutn(&seq_raw, 4);
  msg->seqid = (int32_t)wire::get_u32_be(seq_raw.data());
  source->cutn(&msg->payload, frame_len - 8 - 4 - name_len);
  return ParseResult::make_ok(msg);
}

void ProcessThriftRequest(InputMessageBase* msg_base) {
  ThriftMessage* msg = (ThriftMessage*)msg_base;
  SocketUniquePtr sock;
  if (Socket::Address(msg->socket_id, &sock) != 0) {
    delete msg;
    return;
  }
  Server* server = (Server*)sock->user();
  const MethodFn* fn = server != nullptr ? server->FindMethod("thrift", msg->method) : nullptr;
  SocketId sid = sock->id();
  int32_t seqid = msg->seqid;
  std::string method = msg->method;
  bool oneway = msg->msg_type == T_ONEWAY;
  if (fn == nullptr) {
    if (!oneway) {
      // TApplicationException: struct {1: string message, 2: i32 type}
      IOBuf exc;
      std::string body;
      body.push_back((char)11);  // string field
      char b[4];
      body.push_back(0);
      body.push_back(1);  // field id 1
      std::string text = "unknown thrift method " + method;
      wire::put_u32_be(b, (uint32_t)text.size());
      body.append(b, 4);
      body.append(text);
      body.push_back((char)8);  // i32 field
      body.push_back(0);
      body.push_back(2);
      wire::put_u32_be(b, 1);  // UNKNOWN_METHOD
      body.append(b, 4);
      body.push_back(0);  // stop
      exc.append(body);
      IOBuf out;
      PackThriftMessage(&out, T_EXCEPTION, method, seqid, exc);
      sock->Write(&out);
    }
    delete msg;
    return;
  }
  Controller* cntl = new Controller;
  cntl->server_ = server;
  cntl->server_socket_ = sid;
  IOBuf* resp = new IOBuf;
  Closure* done = NewCallback([sid, seqid, method, cntl, resp, oneway] {
    if (!oneway) {
      SocketUniquePtr s;
      if (Socket::Address(sid, &s) == 0) {
        IOBuf out;
        PackThriftMessage(&out, cntl->Failed() ? T_EXCEPTION : T_REPLY, method, seqid, *resp);
        s->Write(&out);
      }
    }
    delete resp;
    delete cntl;
  });
  (*fn)(cntl, msg->payload, resp, done);
  delete msg;
}

void ProcessThriftResponse(InputMessageBase* msg_base) {
  ThriftMessage* msg = (ThriftMessage*)msg_base;
  SocketUniquePtr sock;
  uint64_t cid = 0;
  if (Socket::Address(msg->socket_id, &sock) == 0) cid = sock->pop_pipeline();
  if (cid == 0) {
    delete msg;
    return;
  }
  void* data = nullptr;
  if (session_lock(cid, &data) != 0) {
    delete msg;
    return;
  }
  Controller* cntl = (Controller*)data;
  if (sock) sock->remove_pending_session(cid);
  if (msg->msg_type == T_EXCEPTION) {
    cntl->SetFailed(ERESPONSE, "thrift exception from server");
  } else if (cntl->call.response != nullptr) {
    cntl->call.response->clear();
    cntl->call.response->append(std::move(msg->payload));
  }
  delete msg;
  EndRPC(cntl, cid);
}

void PackThriftRequest(IOBuf* out, Controller* cntl, uint64_t /*cid*/) {
  static std::atomic<int32_t> seq{1};
  PackThriftMessage(out, T_CALL, cntl->call.method_name,
                    seq.fetch_add(1, std::memory_order_relaxed), cntl->call.request_buf);
}

}  // namespace

void RegisterThriftProtocol() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    Protocol p;
    p.parse = ParseThrift;
    p.process_request = ProcessThriftRequest;
    p.process_response = ProcessThriftResponse;
    p.pack_request = PackThriftRequest;
    p.client_pipelined = true;
    p.support_server = true;
    p.support_client = true;
    p.name = "thrift";
    RegisterProtocol(p);
  });
}

}  // namespace policy
}  // namespace bam
