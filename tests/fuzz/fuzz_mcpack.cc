// Fuzz: mcpack v2 parser + reserializer round-trip.
#include <string>

#include "base/mcpack.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  bam::mcpack::Value v;
  std::string err;
  if (bam::mcpack::Parse((const char*)data, n, &v, &err)) {
    std::string out, json;
    bam::mcpack::Serialize(v, &out);
    bam::mcpack::ToJson(v, &json);
  }
  return 0;
}
