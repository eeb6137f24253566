// Fuzz: HPACK header-block decoder (rpc/policy/hpack.cc — in-tree RFC
// 7541 implementation; parity intent: reference details/hpack.cpp is
// exercised by test/fuzzing/fuzz_hpack.cpp). Also round-trips whatever
// decodes: encode(decode(x)) must decode again to the same headers.
#include <string>
#include <vector>

#include "rpc/policy/hpack.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  bam::hpack::Decoder dec;
  std::vector<bam::hpack::Header> headers;
  if (!dec.Decode((const char*)data, n, &headers)) return 0;
  bam::hpack::Encoder enc;
  std::string wire;
  enc.Encode(headers, &wire);
  bam::hpack::Decoder dec2;
  std::vector<bam::hpack::Header> again;
  if (!dec2.Decode(wire.data(), wire.size(), &again)) __builtin_trap();
  if (again.size() != headers.size()) __builtin_trap();
  for (size_t i = 0; i < headers.size(); ++i) {
    if (again[i].first != headers[i].first || again[i].second != headers[i].second)
      __builtin_trap();
  }
  return 0;
}
