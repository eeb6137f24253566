// brpc_amd: Redis protocol (RESP) — client through Channel (pipelined FIFO
// correlation, parity: reference brpc/redis.h + policy/redis_protocol.cpp)
// and SERVER side (RedisService command handlers, parity:
// brpc/redis.h:194 RedisService/RedisCommandHandler) so a brpc_amd server
// can speak RESP to any redis client.
#pragma once

#include <functional>
#include <map>
#include <string>
#include <vector>

#include "base/iobuf.h"

namespace bam {

struct RedisReply {
  enum Type { NIL, STATUS, ERROR, INTEGER, STRING, ARRAY };
  Type type = NIL;
  std::string str;  // STATUS / ERROR / STRING payload
  int64_t integer = 0;
  std::vector<RedisReply> elements;

  static RedisReply Status(std::string s) {
    RedisReply r;
    r.type = STATUS;
    r.str = std::move(s);
    return r;
  }
  static RedisReply Error(std::string s) {
    RedisReply r;
    r.type = ERROR;
    r.str = std::move(s);
    return r;
  }
  static RedisReply Integer(int64_t v) {
    RedisReply r;
    r.type = INTEGER;
    r.integer = v;
    return r;
  }
  static RedisReply Bulk(std::string s) {
    RedisReply r;
    r.type = STRING;
    r.str = std::move(s);
    return r;
  }
  static RedisReply Nil() { return RedisReply(); }

  void SerializeTo(std::string* out) const;
};

// Incremental RESP parse. Returns bytes consumed (>0), 0 if more data is
// needed, -1 on malformed input.
ssize_t ParseRedisValue(const char* data, size_t n, RedisReply* out);

// Encodes an argv command as a RESP array of bulk strings.
void EncodeRedisCommand(const std::vector<std::string>& args, std::string* out);

// Server-side command dispatch (case-insensitive command names).
class RedisService {
 public:
  typedef std::function<RedisReply(const std::vector<std::string>& args)> CommandHandler;
  void AddCommandHandler(const std::string& command, CommandHandler handler);
  const CommandHandler* FindHandler(const std::string& command) const;

 private:
  std::map<std::string, CommandHandler> handlers_;  // lower-cased keys
};

namespace policy {
void RegisterRedisProtocol();
}

}  // namespace bam
