// Fuzz: snappy decompressor on arbitrary bytes (parity: reference uses
// upstream snappy; ours is a clean-room codec — hip/snappy.hip mirrors it).
#include <string>

#include "base/snappy.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  std::string out;
  bam::snappy::Uncompress((const char*)data, n, &out);
  // Also round-trip compress of the raw input: must always succeed.
  std::string comp, back;
  bam::snappy::Compress((const char*)data, n, &comp);
  if (!bam::snappy::Uncompress(comp.data(), comp.size(), &back) ||
      back != std::string((const char*)data, n)) {
    __builtin_trap();
  }
  return 0;
}
