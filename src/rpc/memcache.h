// brpc_amd: memcached binary protocol client (parity: reference
// brpc/memcache.h + policy/memcache_binary_protocol.cpp).
#pragma once

#include <string>

#include "rpc/channel.h"

namespace bam {

struct MemcacheResponse {
  uint8_t opcode = 0;
  uint16_t status = 0;  // 0 = ok, 1 = key not found, ...
  uint64_t cas = 0;
  std::string extras;
  std::string key;
  std::string value;
};

void PackMemcacheRequest(IOBuf* out, uint8_t opcode, const std::string& key,
                         const std::string& value, const std::string& extras, uint64_t cas,
                         uint16_t vbucket = 0);

class MemcacheClient {
 public:
  // Uses the pipelined memcache protocol over a shared connection.
  explicit MemcacheClient(const std::string& addr, int timeout_ms = 1000);
  bool ok() const { return init_ok_; }

  // All return 0 on success; >10000 = memcache status + 10000;
  // rpc errors otherwise.
  int Set(const std::string& key, const std::string& value, uint32_t flags = 0,
          uint32_t exptime = 0);
  int Get(const std::string& key, std::string* value);
  int Delete(const std::string& key);
  int Version(std::string* version);
  int RawCall(uint8_t opcode, const std::string& key, const std::string& value,
              const std::string& extras, MemcacheResponse* out, uint16_t vbucket = 0);

  // Couchbase parity (reference policy/couchbase_authenticator.cpp +
  // couchbase_protocol.cpp): SASL PLAIN authentication over the memcache
  // binary protocol (opcode 0x21, mechanism "PLAIN",
  // value = "\0user\0password"). 0 on success.
  int SaslAuthPlain(const std::string& user, const std::string& password);

 private:
  Channel channel_;
  bool init_ok_ = false;
};

namespace policy {
void RegisterMemcacheProtocol();
}

}  // namespace bam
