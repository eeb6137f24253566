// brpc_amd: Snappy compression (clean-room implementation of the public
// snappy format: varint preamble + literal/copy tag stream, 64 KiB match
// window). Capability parity: reference butil/third_party/snappy (used by
// policy/snappy_compress.cpp). The gfx950 block-parallel variant lives in
// hip/snappy.hip; this host codec is its reference oracle.
#pragma once

#include <stddef.h>
#include <stdint.h>

#include <string>

namespace bam {
namespace snappy {

size_t MaxCompressedLength(size_t source_len);

// Returns compressed size written to dst (must have MaxCompressedLength room).
size_t RawCompress(const char* src, size_t n, char* dst);

// Returns false on corrupt input. *uncompressed_len from the preamble.
bool GetUncompressedLength(const char* compressed, size_t n, size_t* result);
bool RawUncompress(const char* compressed, size_t n, char* dst);

// Convenience.
void Compress(const char* src, size_t n, std::string* out);
bool Uncompress(const char* compressed, size_t n, std::string* out);

}  // namespace snappy
}  // namespace bam
