// Fuzz: in-tree HTTP/2 session state machine (rpc/policy/h2_session.cc):
// arbitrary bytes after the client preface drive the server-side frame
// parser (SETTINGS/HEADERS+HPACK/DATA/WINDOW_UPDATE/CONTINUATION/RST/
// GOAWAY, flow-control accounting). Parity intent: the reference fuzzes
// its h2 parser via test/fuzzing/fuzz_http (h2 upgrade path).
#include <string>
#include <vector>

#include "rpc/policy/h2_session.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  bam::policy::H2Session::Callbacks cbs;
  size_t sink = 0;
  cbs.on_header = [&](int32_t, const std::string& k, const std::string& v) {
    sink += k.size() + v.size();
  };
  cbs.on_data = [&](int32_t, const char*, size_t len) { sink += len; };
  cbs.on_end_stream = [&](int32_t sid) { sink += (size_t)sid; };
  cbs.on_rst = [&](int32_t, uint32_t) { ++sink; };
  cbs.on_goaway = [&](uint32_t) { ++sink; };
  bam::policy::H2Session server(true, cbs);
  // Valid preface so fuzz bytes reach the frame parser, then the input.
  static const char kPreface[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";
  std::string in(kPreface, sizeof(kPreface) - 1);
  in.append((const char*)data, n);
  size_t off = 0;
  // Feed in two chunks to exercise partial-frame buffering.
  for (int round = 0; round < 2 && off < in.size(); ++round) {
    size_t take = round == 0 ? in.size() / 2 + 1 : in.size() - off;
    ssize_t c = server.Consume(in.data() + off, take);
    if (c < 0) break;
    off += (size_t)c;
    std::string out;
    server.TakeOutput(&out);  // drain acks/window updates
  }
  (void)sink;
  return 0;
}
