// brpc_amd: couchbase client — vbucket-aware routing over the memcache
// binary protocol (parity: reference policy/couchbase_protocol.cpp +
// couchbase_authenticator; round 1 stopped at SASL PLAIN).
// Cluster topology comes from the REST bucket config
// (/pools/default/b/<bucket> → vBucketServerMap); keys map to vbuckets by
// CRC32(key) >> 16 & (n-1); requests carry the vbucket id in the binary
// header and NOT_MY_VBUCKET (0x0007) triggers a config refetch + retry.
#pragma once

#include <map>
#include <memory>
#include <string>
#include <vector>

#include "rpc/memcache.h"

namespace bam {

class CouchbaseClient {
 public:
  // config_addr: host:port of the REST config endpoint; bucket: name.
  // Optional SASL PLAIN credentials for the data nodes.
  int Init(const std::string& config_addr, const std::string& bucket,
           const std::string& user = "", const std::string& password = "");

  int Set(const std::string& key, const std::string& value);
  int Get(const std::string& key, std::string* value);
  int Delete(const std::string& key);

  // key -> vbucket id (CRC32 of the key, high half, masked).
  static uint16_t VBucketOf(const std::string& key, size_t nvbuckets);

  int nvbuckets() const { return (int)vbucket_primary_.size(); }
  int nservers() const { return (int)servers_.size(); }
  const std::string& last_error() const { return err_; }

 private:
  int FetchConfig();
  MemcacheClient* conn_for(int server_idx);
  int Op(uint8_t opcode, const std::string& key, const std::string& value,
         const std::string& extras, MemcacheResponse* out);

  std::string config_addr_, bucket_, user_, password_;
  std::vector<std::string> servers_;   // "host:port" (data port)
  std::vector<int> vbucket_primary_;   // vbucket -> server index
  std::map<int, std::unique_ptr<MemcacheClient>> conns_;
  std::string err_;
};

}  // namespace bam
