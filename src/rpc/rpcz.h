// brpc_amd: rpcz — per-RPC span tracing (parity: reference brpc/span.h +
// builtin/rpcz_service.cpp). Spans are recorded into a bounded in-memory
// ring (the reference uses a leveldb-backed SpanDB; ours is a ring buffer
// sized by -rpcz_max_spans) and rendered at the /rpcz builtin page.
// trace/log ids propagate through the std protocol meta.
#pragma once

#include <stdint.h>

#include <string>

#include "base/endpoint.h"
#include "base/iobuf.h"

namespace bam {
namespace rpcz {

struct Span {
  int64_t start_us = 0;
  int64_t end_us = 0;
  std::string full_method;
  EndPoint remote;
  int error_code = 0;
  uint64_t log_id = 0;
  bool server_side = false;
  uint64_t request_size = 0;
  uint64_t response_size = 0;
};

bool enabled();
void set_enabled(bool on);
void RecordSpan(const Span& span);
void DumpRecentSpans(IOBuf* out, bool verbose);
// Reads back the newest `max` spans from the recordio SpanDB
// (-rpcz_db_path); header-only output when no db is configured.
void DumpPersistedSpans(IOBuf* out, int max);
int64_t span_count();

}  // namespace rpcz
}  // namespace bam
