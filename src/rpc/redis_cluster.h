// brpc_amd: redis cluster client — slot-aware routing with MOVED/ASK
// redirect handling (parity: reference redis_cluster client paths in
// brpc/redis*.cpp). Topology from CLUSTER SLOTS; keys map to slots by
// CRC16-CCITT(key) % 16384 honoring {hash tags}.
#pragma once

#include <map>
#include <string>
#include <vector>

#include "rpc/redis.h"

namespace bam {

class RedisClusterClient {
 public:
  // seed: "host:port" of any cluster node.
  int Init(const std::string& seed, int timeout_ms = 1000);

  // Runs one command; the routing key is args[1] (redis convention).
  // Follows -MOVED (slot map update + retry) and -ASK (one-shot redirect
  // with ASKING). 0 on success.
  int Command(const std::vector<std::string>& args, RedisReply* out);

  static uint16_t SlotOf(const std::string& key);  // CRC16 % 16384, hash tags
  int nslots_mapped() const;
  const std::string& last_error() const { return err_; }

 private:
  int RefreshSlots(const std::string& via);
  int CallNode(const std::string& addr, const std::vector<std::string>& args,
               RedisReply* out, bool asking);

  struct Range {
    int start, end;
    std::string addr;
  };
  std::vector<Range> ranges_;
  std::string seed_;
  int timeout_ms_ = 1000;
  std::string err_;
};

}  // namespace bam
