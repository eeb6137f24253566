// brpc_amd: mongo wire protocol, server side.
// Parity: reference policy/mongo_protocol.cpp + mongo_head.h +
// mongo_service_adaptor.h (clean-room): a 16-byte little-endian header
// {message_length, request_id, response_to, op_code}; the op_code doubles
// as the magic number (legacy set 1/1000/2001/2002/2004/2005/2006/2007 plus
// modern OP_MSG 2013). Like the reference, the framework does NOT
// interpret BSON — the raw body goes to the user's mongo handler
// (ServerOptions::mongo_handler ≙ MongoServiceAdaptor), which fills a
// reply; OP_QUERY gets an OP_REPLY envelope, OP_MSG gets an OP_MSG one.
// Per-connection session state rides Socket::protocol_ctx
// (≙ MongoContext on parsing_context).
#include <string.h>

#include <mutex>
#include <string>

#include "base/iobuf.h"
#include "base/logging.h"
#include "rpc/policy/http_protocol.h"
#include "rpc/protocol.h"
#include "rpc/server.h"
#include "rpc/socket.h"

namespace bam {
namespace policy {

namespace {

constexpr int32_t kOpReply = 1;
constexpr int32_t kOpDbMsg = 1000;
constexpr int32_t kOpUpdate = 2001;
constexpr int32_t kOpInsert = 2002;
constexpr int32_t kOpQuery = 2004;
constexpr int32_t kOpGetMore = 2005;
constexpr int32_t kOpDelete = 2006;
constexpr int32_t kOpKillCursors = 2007;
constexpr int32_t kOpMsg = 2013;

bool is_mongo_opcode(int32_t op) {
  switch (op) {
    case kOpReply:
    case kOpDbMsg:
    case kOpUpdate:
    case kOpInsert:
    case kOpQuery:
    case kOpGetMore:
    case kOpDelete:
    case kOpKillCursors:
    case kOpMsg:
      return true;
    default:
      return false;
  }
}

inline int32_t get_i32_le(const char* p) {
  int32_t v;
  memcpy(&v, p, 4);
  return v;
}
inline void put_i32_le(std::string* out, int32_t v) {
  char b[4];
  memcpy(b, &v, 4);
  out->append(b, 4);
}
inline void put_i64_le(std::string* out, int64_t v) {
  char b[8];
  memcpy(b, &v, 8);
  out->append(b, 8);
}

struct MongoMsg : public InputMessageBase {
  MongoHeader head;
  IOBuf body;  // everything after the 16-byte header
};

ParseResult ParseMongo(IOBuf* source, Socket* sock, bool) {
  // Only sockets owned by a Server with a mongo handler speak mongo
  // (parity: reference checks options().mongo_service_adaptor first).
  Server* server = (Server*)sock->user();
  if (server == nullptr || !server->options().mongo_handler)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  char aux[16];
  if (source->size() < 16) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(aux, 16);
  int32_t msg_len = get_i32_le(h);
  int32_t op = get_i32_le(h + 12);
  if (!is_mongo_opcode(op) || msg_len < 16 || msg_len > (64 << 20))
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  if (source->size() < (size_t)msg_len)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  MongoMsg* msg = new MongoMsg;
  msg->head.message_length = msg_len;
  msg->head.request_id = get_i32_le(h + 4);
  msg->head.response_to = get_i32_le(h + 8);
  msg->head.op_code = op;
  source->pop_front(16);
  source->cutn(&msg->body, (size_t)msg_len - 16);
  return ParseResult::make_ok(msg);
}

void ProcessMongoRequest(InputMessageBase* mb) {
  MongoMsg* msg = (MongoMsg*)mb;
  SocketUniquePtr sock;
  if (Socket::Address(msg->socket_id, &sock) != 0) {
    delete msg;
    return;
  }
  Server* server = (Server*)sock->user();
  if (server == nullptr || !server->options().mongo_handler) {
    delete msg;
    return;
  }
  MongoReply reply;
  server->options().mongo_handler(msg->head, msg->body, &reply);
  server->nprocessed.fetch_add(1, std::memory_order_relaxed);
  if (reply.body.empty() && !reply.always_reply) {
    delete msg;  // fire-and-forget ops (legacy INSERT/UPDATE/DELETE)
    return;
  }
  static std::atomic<int32_t> g_reply_id{1};
  std::string out;
  if (msg->head.op_code == kOpMsg) {
    // OP_MSG reply: header + flagBits(0) + section payload from handler
    put_i32_le(&out, (int32_t)(16 + 4 + reply.body.size()));
    put_i32_le(&out, g_reply_id.fetch_add(1, std::memory_order_relaxed));
    put_i32_le(&out, msg->head.request_id);
    put_i32_le(&out, kOpMsg);
    put_i32_le(&out, 0);  // flagBits
  } else {
    // Legacy OP_REPLY envelope: responseFlags, cursorId, startingFrom,
    // numberReturned, then the handler's documents.
    put_i32_le(&out, (int32_t)(16 + 20 + reply.body.size()));
    put_i32_le(&out, g_reply_id.fetch_add(1, std::memory_order_relaxed));
    put_i32_le(&out, msg->head.request_id);
    put_i32_le(&out, kOpReply);
    put_i32_le(&out, reply.response_flags);
    put_i64_le(&out, reply.cursor_id);
    put_i32_le(&out, 0);  // startingFrom
    put_i32_le(&out, reply.number_returned);
  }
  IOBuf packet;
  packet.append(out);
  packet.append(reply.body);
  sock->Write(&packet);
  delete msg;
}

}  // namespace

void RegisterMongoProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParseMongo;
    p.process_request = ProcessMongoRequest;
    p.support_server = true;
    p.support_client = false;
    p.name = "mongo";
    RegisterProtocol(p);
  });
}

}  // namespace policy
}  // namespace bam
