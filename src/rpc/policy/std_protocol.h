// brpc_amd: the "std" protocol — wire-compatible with the reference's
// baidu_std (brpc/policy/baidu_rpc_protocol.cpp): 12-byte header
// "PRPC" + body_size + meta_size, protobuf-encoded RpcMeta, then payload
// [+ attachment]. Meta encode/decode is the hand-rolled codec in
// rpc/wire.h (and the gfx950 kernel variant in hip/meta_codec.hip).
#pragma once

#include "base/iobuf.h"
#include "fiber/session.h"

namespace bam {

class Controller;

namespace policy {

// Parsed RpcMeta (subset used by the std protocol).
struct RpcMeta {
  // request
  std::string service_name;
  std::string method_name;
  uint64_t log_id = 0;
  // trace propagation (≙ reference baidu_rpc_meta.proto RpcRequestMeta
  // fields 4-6: trace_id/span_id/parent_span_id)
  uint64_t trace_id = 0;
  uint64_t span_id = 0;
  uint64_t parent_span_id = 0;
  // response
  int error_code = 0;
  std::string error_text;
  bool has_response = false;
  bool has_request = false;
  // common
  int compress_type = 0;
  int64_t correlation_id = 0;
  int32_t attachment_size = 0;
  std::string auth_data;   // authentication_data (field 7, first request on a connection)
  uint64_t stream_id = 0;  // StreamSettings (field 8 sub-message)
};

void SerializeRpcMeta(const RpcMeta& meta, std::string* out);
bool ParseRpcMeta(const char* data, size_t n, RpcMeta* out);

// Builds header+meta+payload(+attachment) for a client request.
void PackStdRequest(IOBuf* out, Controller* cntl, SessionId correlation_id);

// Registers the protocol once (idempotent).
void RegisterStdProtocol();

}  // namespace policy
}  // namespace bam
