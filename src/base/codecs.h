// brpc_amd: small codecs — base64, SHA-1, MurmurHash3 (parity: reference
// butil/base64.cc, sha1_portable.cc, third_party/murmurhash3).
#pragma once

#include <stdint.h>

#include <string>

namespace bam {

void Base64Encode(const std::string& input, std::string* output);
bool Base64Decode(const std::string& input, std::string* output);

// 20-byte binary digest.
std::string SHA1Hash(const std::string& input);
std::string SHA1HexDigest(const std::string& input);

uint32_t MurmurHash3_32(const void* key, size_t len, uint32_t seed);

}  // namespace bam
