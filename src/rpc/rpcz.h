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
  uint64_t trace_id = 0;
  uint64_t span_id = 0;
  uint64_t parent_span_id = 0;
  int64_t end_us = 0;
  std::string full_method;
  EndPoint remote;
  int error_code = 0;
  uint64_t log_id = 0;
  bool server_side = false;
  uint64_t request_size = 0;
  uint64_t response_size = 0;
};

// Ambient trace for the CURRENT execution context (fiber-local; plain
// TLS off-fiber): servers export the inbound ids while the handler runs
// so nested client calls chain parent_span_id (≙ reference span.h:153
// TLS parent chaining via bthread_set_span_funcs).
struct TraceContext {
  uint64_t trace_id = 0;
  uint64_t span_id = 0;
};
TraceContext current_trace();
void set_current_trace(uint64_t trace_id, uint64_t span_id);
void clear_current_trace();

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
