#include <algorithm>
#include <mutex>

#include "base/logging.h"
#include "fiber/session.h"
#include "rpc/channel.h"
#include "rpc/controller.h"
#include "rpc/redis.h"
#include "rpc/server.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

// ---------------- RESP codec ----------------

void RedisReply::SerializeTo(std::string* out) const {
  switch (type) {
    case STATUS:
      out->push_back('+');
      out->append(str);
      out->append("\r\n");
      break;
    case ERROR:
      out->push_back('-');
      out->append(str);
      out->append("\r\n");
      break;
    case INTEGER:
      out->push_back(':');
      out->append(std::to_string(integer));
      out->append("\r\n");
      break;
    case STRING:
      out->push_back('$');
      out->append(std::to_string(str.size()));
      out->append("\r\n");
      out->append(str);
      out->append("\r\n");
      break;
    case NIL:
      out->append("$-1\r\n");
      break;
    case ARRAY:
      out->push_back('*');
      out->append(std::to_string(elements.size()));
      out->append("\r\n");
      for (const RedisReply& e : elements) e.SerializeTo(out);
      break;
  }
}

namespace {
// returns end-of-line index (of \r) or npos
size_t find_crlf(const char* data, size_t n, size_t from) {
  for (size_t i = from; i + 1 < n; ++i) {
    if (data[i] == '\r' && data[i + 1] == '\n') return i;
  }
  return (size_t)-1;
}
}  // namespace

ssize_t ParseRedisValue(const char* data, size_t n, RedisReply* out) {
  if (n == 0) return 0;
  char t = data[0];
  size_t eol = find_crlf(data, n, 1);
  if (eol == (size_t)-1) return n > 64 * 1024 && (t == '+' || t == '-' || t == ':') ? -1 : 0;
  std::string line(data + 1, eol - 1);
  size_t consumed = eol + 2;
  switch (t) {
    case '+':
      *out = RedisReply::Status(line);
      return (ssize_t)consumed;
    case '-':
      *out = RedisReply::Error(line);
      return (ssize_t)consumed;
    case ':':
      *out = RedisReply::Integer(strtoll(line.c_str(), nullptr, 10));
      return (ssize_t)consumed;
    case '$': {
      long len = strtol(line.c_str(), nullptr, 10);
      if (len < 0) {
        *out = RedisReply::Nil();
        return (ssize_t)consumed;
      }
      if (n < consumed + (size_t)len + 2) return 0;
      *out = RedisReply::Bulk(std::string(data + consumed, (size_t)len));
      return (ssize_t)(consumed + len + 2);
    }
    case '*': {
      long count = strtol(line.c_str(), nullptr, 10);
      RedisReply arr;
      arr.type = RedisReply::ARRAY;
      if (count < 0) {
        *out = RedisReply::Nil();
        return (ssize_t)consumed;
      }
      size_t pos = consumed;
      for (long i = 0; i < count; ++i) {
        RedisReply elem;
        ssize_t c = ParseRedisValue(data + pos, n - pos, &elem);
        if (c <= 0) return c;
        arr.elements.push_back(std::move(elem));
        pos += (size_t)c;
      }
      *out = std::move(arr);
      return (ssize_t)pos;
    }
    default:
      return -1;
  }
}

void EncodeRedisCommand(const std::vector<std::string>& args, std::string* out) {
  out->push_back('*');
  out->append(std::to_string(args.size()));
  out->append("\r\n");
  for (const std::string& a : args) {
    out->push_back('$');
    out->append(std::to_string(a.size()));
    out->append("\r\n");
    out->append(a);
    out->append("\r\n");
  }
}

// ---------------- RedisService ----------------

void RedisService::AddCommandHandler(const std::string& command, CommandHandler handler) {
  std::string key = command;
  std::transform(key.begin(), key.end(), key.begin(), ::tolower);
  handlers_[key] = std::move(handler);
}

const RedisService::CommandHandler* RedisService::FindHandler(const std::string& command) const {
  std::string key = command;
  std::transform(key.begin(), key.end(), key.begin(), ::tolower);
  auto it = handlers_.find(key);
  return it == handlers_.end() ? nullptr : &it->second;
}

// ---------------- protocol glue ----------------

namespace policy {

namespace {

struct RedisMessage : public InputMessageBase {
  RedisReply value;
};

ParseResult ParseRedis(IOBuf* source, Socket* sock, bool /*eof*/) {
  char probe;
  if (source->size() < 1) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* p = (const char*)source->fetch(&probe, 1);
  const bool server_side = sock->user() != nullptr;
  // Server sees inline or array commands ('*'); client sees any RESP type.
  if (server_side) {
    if (*p != '*') return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
    Server* srv = (Server*)sock->user();
    if (srv->redis_service() == nullptr)
      return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  } else {
    if (*p != '+' && *p != '-' && *p != ':' && *p != '$' && *p != '*')
      return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
    // only treat as redis when the socket already expects redis replies
    if (sock->preferred_protocol_index < 0) {
      // no pipelined request outstanding: cannot be our redis reply
    }
  }
  std::string flat;
  size_t scan = std::min<size_t>(source->size(), 16u << 20);
  source->copy_to(&flat, scan, 0);
  RedisMessage* msg = new RedisMessage;
  ssize_t consumed = ParseRedisValue(flat.data(), flat.size(), &msg->value);
  if (consumed == 0) {
    delete msg;
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  }
  if (consumed < 0) {
    delete msg;
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  }
  source->pop_front((size_t)consumed);
  return ParseResult::make_ok(msg);
}

void ProcessRedisRequest(InputMessageBase* msg_base) {
  RedisMessage* msg = (RedisMessage*)msg_base;
  SocketUniquePtr sock;
  if (Socket::Address(msg->socket_id, &sock) != 0) {
    delete msg;
    return;
  }
  Server* srv = (Server*)sock->user();
  RedisService* service = srv != nullptr ? srv->redis_service() : nullptr;
  RedisReply reply;
  if (service == nullptr || msg->value.type != RedisReply::ARRAY ||
      msg->value.elements.empty()) {
    reply = RedisReply::Error("ERR bad command");
  } else {
    std::vector<std::string> args;
    for (const RedisReply& e : msg->value.elements) args.push_back(e.str);
    const RedisService::CommandHandler* h = service->FindHandler(args[0]);
    if (h == nullptr) {
      reply = RedisReply::Error("ERR unknown command '" + args[0] + "'");
    } else {
      reply = (*h)(args);
    }
  }
  std::string out;
  reply.SerializeTo(&out);
  IOBuf out_buf;
  out_buf.append(out);
  sock->Write(&out_buf);
  if (srv != nullptr) srv->nprocessed.fetch_add(1, std::memory_order_relaxed);
  delete msg;
}

void ProcessRedisResponse(InputMessageBase* msg_base) {
  RedisMessage* msg = (RedisMessage*)msg_base;
  SocketUniquePtr sock;
  uint64_t cid = 0;
  if (Socket::Address(msg->socket_id, &sock) == 0) cid = sock->pop_pipeline();
  if (cid == 0) {
    delete msg;
    return;
  }
  void* data = nullptr;
  if (session_lock(cid, &data) != 0) {
    delete msg;  // timed out already
    return;
  }
  Controller* cntl = (Controller*)data;
  if (sock) sock->remove_pending_session(cid);
  if (msg->value.type == RedisReply::ERROR) {
    cntl->SetFailed(ERESPONSE, msg->value.str);
  } else if (cntl->call.response != nullptr) {
    std::string raw;
    msg->value.SerializeTo(&raw);
    cntl->call.response->clear();
    cntl->call.response->append(raw);
  }
  delete msg;
  EndRPC(cntl, cid);
}

void PackRedisRequest(IOBuf* out, Controller* cntl, uint64_t /*cid*/) {
  out->append(cntl->call.request_buf);  // already RESP-encoded by the caller
}

}  // namespace

void RegisterRedisProtocol() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    Protocol p;
    p.parse = ParseRedis;
    p.process_request = ProcessRedisRequest;
    p.process_response = ProcessRedisResponse;
    p.pack_request = PackRedisRequest;
    p.client_pipelined = true;
    p.support_server = true;
    p.support_client = true;
    p.name = "redis";
    RegisterProtocol(p);
  });
}

}  // namespace policy
}  // namespace bam
