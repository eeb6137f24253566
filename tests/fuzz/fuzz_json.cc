// Fuzz: JSON parser + reserializer (parity: reference
// test/fuzzing/fuzz_json.cpp).
#include <string>

#include "base/json.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  bam::json::Value v;
  std::string err;
  if (bam::json::Parse(std::string((const char*)data, n), &v, &err)) {
    std::string out;
    bam::json::Serialize(v, &out);
  }
  return 0;
}
