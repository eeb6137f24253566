// Fuzz: IOBuf::cut_until — delimiter split from input, scanned across an
// artificial block boundary; result must equal a straightforward
// std::string::find on the same bytes.
#include <string>

#include "base/iobuf.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  if (n < 2) return 0;
  size_t dlen = 1 + data[0] % 4;
  if (n < 1 + dlen) return 0;
  std::string delim((const char*)data + 1, dlen);
  std::string body((const char*)data + 1 + dlen, n - 1 - dlen);
  bam::IOBuf buf;
  size_t half = body.size() / 2;
  buf.append(body.data(), half);
  bam::IOBuf second;  // force a separate block for the tail
  second.append(body.data() + half, body.size() - half);
  buf.append(second);
  bam::IOBuf out;
  int rc = buf.cut_until(&out, delim);
  size_t pos = body.find(delim);
  if (pos == std::string::npos) {
    if (rc == 0) __builtin_trap();
  } else {
    if (rc != 0 || out.to_string() != body.substr(0, pos)) __builtin_trap();
    std::string rest = buf.to_string();
    if (rest != body.substr(pos + delim.size())) __builtin_trap();
  }
  return 0;
}
