// brpc_amd: pluggable compression registry.
// Parity: reference brpc/compress.h (CompressHandler registered per
// CompressType in global.cpp:418-426): snappy (base/snappy.cc host codec;
// gfx950 kernel in hip/snappy.hip picked when payload blocks are
// HBM-resident) and gzip (zlib).
#pragma once

#include "base/iobuf.h"
#include "rpc/controller.h"

namespace bam {

struct CompressHandler {
  bool (*Compress)(const IOBuf& in, IOBuf* out);
  bool (*Decompress)(const IOBuf& in, IOBuf* out);
  const char* name;
};

int RegisterCompressHandler(CompressType type, CompressHandler handler);
const CompressHandler* FindCompressHandler(CompressType type);

// Convenience; returns false if no handler / codec failure.
bool ApplyCompress(CompressType type, const IOBuf& in, IOBuf* out);
bool ApplyDecompress(CompressType type, const IOBuf& in, IOBuf* out);

// Registers snappy + gzip (idempotent; called from protocol init).
void RegisterBuiltinCompressHandlers();

}  // namespace bam
