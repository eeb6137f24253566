// Fuzz: HTTP/1.1 request-head parser (parity: reference
// test/fuzzing/fuzz_http.cpp).
#include <string>

#include "rpc/policy/http_protocol.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  bam::policy::HttpRequest req;
  bam::policy::ParseHttpHead(std::string((const char*)data, n), &req);
  return 0;
}
