// brpc_amd: runtime support for tools/bamproto.py generated code.
// Generated message structs call these helpers for wire encode/decode;
// generated service bases / stubs bind typed methods onto Server/Channel
// (parity intent: protoc-generated google::protobuf stubs integrating
// with reference brpc/channel.h:189-228 + server.cpp:844-875).
#pragma once

#include <stdint.h>
#include <string.h>

#include <string>
#include <vector>

namespace bam {
namespace pbgen {

inline void put_varint(std::string* out, uint64_t v) {
  while (v >= 0x80) {
    out->push_back((char)(v | 0x80));
    v >>= 7;
  }
  out->push_back((char)v);
}

inline bool get_varint(const char*& p, const char* end, uint64_t* v) {
  *v = 0;
  int shift = 0;
  while (p < end && shift < 64) {
    uint8_t b = (uint8_t)*p++;
    *v |= (uint64_t)(b & 0x7f) << shift;
    if ((b & 0x80) == 0) return true;
    shift += 7;
  }
  return false;
}

inline uint64_t zigzag_enc(int64_t v) { return ((uint64_t)v << 1) ^ (uint64_t)(v >> 63); }
inline int64_t zigzag_dec(uint64_t v) { return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); }

inline void put_tag(std::string* out, int field, int wt) {
  put_varint(out, (uint64_t)(field << 3 | wt));
}

inline void put_len_delim(std::string* out, int field, const std::string& s) {
  put_tag(out, field, 2);
  put_varint(out, s.size());
  out->append(s);
}

inline void put_fixed64(std::string* out, uint64_t v) {
  for (int i = 0; i < 8; ++i) out->push_back((char)(v >> (8 * i)));
}
inline void put_fixed32(std::string* out, uint32_t v) {
  for (int i = 0; i < 4; ++i) out->push_back((char)(v >> (8 * i)));
}
inline void put_double(std::string* out, double d) {
  uint64_t bits;
  memcpy(&bits, &d, 8);
  put_fixed64(out, bits);
}
inline void put_float(std::string* out, float f) {
  uint32_t bits;
  memcpy(&bits, &f, 4);
  put_fixed32(out, bits);
}

inline bool get_fixed64(const char*& p, const char* end, uint64_t* v) {
  if (end - p < 8) return false;
  *v = 0;
  for (int i = 0; i < 8; ++i) *v |= (uint64_t)(uint8_t)p[i] << (8 * i);
  p += 8;
  return true;
}
inline bool get_fixed32(const char*& p, const char* end, uint32_t* v) {
  if (end - p < 4) return false;
  *v = 0;
  for (int i = 0; i < 4; ++i) *v |= (uint32_t)(uint8_t)p[i] << (8 * i);
  p += 4;
  return true;
}
inline bool get_len_delim(const char*& p, const char* end, std::string* s) {
  uint64_t n;
  if (!get_varint(p, end, &n) || (uint64_t)(end - p) < n) return false;
  s->assign(p, (size_t)n);
  p += n;
  return true;
}

// Skips one field of wire type wt; appends the raw bytes (tag already
// consumed by caller, who passes tag_start) to *unknown.
inline bool skip_field(const char*& p, const char* end, int wt) {
  switch (wt) {
    case 0: {
      uint64_t d;
      return get_varint(p, end, &d);
    }
    case 1:
      if (end - p < 8) return false;
      p += 8;
      return true;
    case 2: {
      uint64_t n;
      if (!get_varint(p, end, &n) || (uint64_t)(end - p) < n) return false;
      p += n;
      return true;
    }
    case 5:
      if (end - p < 4) return false;
      p += 4;
      return true;
    default:
      return false;
  }
}

}  // namespace pbgen
}  // namespace bam
