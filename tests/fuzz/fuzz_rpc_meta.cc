// Fuzz: baidu_std RpcMeta codec (parity: reference test/fuzzing/fuzz_butil
// family — SURVEY §4 fuzz targets). Parse arbitrary bytes, then re-serialize
// whatever parsed: must never crash or leak.
#include <string>

#include "rpc/policy/std_protocol.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  bam::policy::RpcMeta meta;
  if (bam::policy::ParseRpcMeta((const char*)data, n, &meta)) {
    std::string out;
    bam::policy::SerializeRpcMeta(meta, &out);
  }
  return 0;
}
