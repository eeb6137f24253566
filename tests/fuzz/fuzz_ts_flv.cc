// Fuzz: FLV parser + MPEG-TS muxer (rpc/flv.cc parse, rpc/ts.cc mux):
// arbitrary bytes through flv::Parse; whatever parses is muxed to TS,
// which must stay 188-byte aligned.
#include <string>
#include <vector>

#include "rpc/flv.h"
#include "rpc/ts.h"

extern "C" int LLVMFuzzerTestOneInput(const unsigned char* data, size_t n) {
  std::vector<bam::flv::Tag> tags;
  if (!bam::flv::Parse(std::string((const char*)data, n), &tags)) return 0;
  std::string out;
  bam::ts::FlvToTs(tags, &out);
  if (out.size() % 188 != 0) __builtin_trap();
  return 0;
}
