// brpc_amd: rpc_dump — sampled capture of live server traffic to recordio
// files, replayable by tools/rpc_replay.py.
// Parity: reference brpc/rpc_dump.h + tools/rpc_replay.
#pragma once

#include <string>

#include "base/iobuf.h"

namespace bam {
namespace rpc_dump {

// Serialized sample record (protobuf-wire: 1=service, 2=method, 3=body).
void EncodeSample(const std::string& service, const std::string& method, const IOBuf& body,
                  std::string* out);
bool DecodeSample(const std::string& rec, std::string* service, std::string* method,
                  std::string* body);

// Called on the server request path; samples per -rpc_dump_ratio when
// -rpc_dump is on, appending to -rpc_dump_file.
void SampleRequest(const std::string& service, const std::string& method, const IOBuf& body);

int64_t sampled_count();

}  // namespace rpc_dump
}  // namespace bam
