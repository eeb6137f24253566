// Fuzz: RESP value parser (parity: reference test/fuzzing/fuzz_redis.cpp).
#include <sys/types.h>

#include "rpc/redis.h"

namespace bam {
ssize_t ParseRedisValue(const char* data, size_t n, RedisReply* out);
}

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  bam::RedisReply reply;
  ssize_t consumed = bam::ParseRedisValue((const char*)data, n, &reply);
  if (consumed > 0) {
    std::string round;
    reply.SerializeTo(&round);
  }
  return 0;
}
