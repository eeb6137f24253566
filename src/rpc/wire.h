// brpc_amd: protobuf wire-format primitives (varint / length-delimited),
// hand-rolled for the fixed RpcMeta schema of the std protocol — no
// libprotobuf dependency. This codec is also the host reference for the
// gfx950 meta-codec kernel (hip/meta_codec.hip) which packs/parses RpcMeta
// batches on-device.
// Wire compatibility target: reference brpc/policy/baidu_rpc_meta.proto.
#pragma once

#include <stdint.h>
#include <string.h>

#include <string>

namespace bam {
namespace wire {

inline void put_varint(std::string* out, uint64_t v) {
  while (v >= 0x80) {
    out->push_back((char)(v | 0x80));
    v >>= 7;
  }
  out->push_back((char)v);
}

inline void put_tag(std::string* out, int field, int wtype) {
  put_varint(out, (uint64_t)(field << 3 | wtype));
}

inline void put_str_field(std::string* out, int field, const std::string& s) {
  put_tag(out, field, 2);
  put_varint(out, s.size());
  out->append(s);
}

inline void put_int_field(std::string* out, int field, int64_t v) {
  put_tag(out, field, 0);
  put_varint(out, (uint64_t)v);
}

inline void put_msg_field(std::string* out, int field, const std::string& sub) {
  put_tag(out, field, 2);
  put_varint(out, sub.size());
  out->append(sub);
}

class Reader {
 public:
  Reader(const char* data, size_t n) : p_(data), end_(data + n), ok_(true) {}

  bool ok() const { return ok_; }
  bool done() const { return p_ >= end_; }

  uint64_t varint() {
    uint64_t v = 0;
    int shift = 0;
    while (p_ < end_) {
      uint8_t b = (uint8_t)*p_++;
      v |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
      if (shift >= 64) break;
    }
    ok_ = false;
    return 0;
  }

  // Returns field number; wtype in *wtype. 0 when done/error.
  int read_tag(int* wtype) {
    if (done()) return 0;
    uint64_t t = varint();
    if (!ok_) return 0;
    *wtype = (int)(t & 7);
    return (int)(t >> 3);
  }

  std::string read_string() {
    uint64_t n = varint();
    if (!ok_ || (uint64_t)(end_ - p_) < n) {
      ok_ = false;
      return std::string();
    }
    std::string s(p_, n);
    p_ += n;
    return s;
  }

  // Reads n raw bytes (fixed32/fixed64 payloads). Empty string on underrun.
  std::string read_fixed(size_t n) {
    if ((size_t)(end_ - p_) < n) {
      ok_ = false;
      return std::string();
    }
    std::string s(p_, n);
    p_ += n;
    return s;
  }

  void skip(int wtype) {
    switch (wtype) {
      case 0:
        varint();
        break;
      case 1:
        p_ += 8;
        break;
      case 2: {
        uint64_t n = varint();
        if ((uint64_t)(end_ - p_) < n) {
          ok_ = false;
          return;
        }
        p_ += n;
        break;
      }
      case 5:
        p_ += 4;
        break;
      default:
        ok_ = false;
    }
    if (p_ > end_) ok_ = false;
  }

 private:
  const char* p_;
  const char* end_;
  bool ok_;
};

inline void put_u32_be(char* p, uint32_t v) {
  p[0] = (char)(v >> 24);
  p[1] = (char)(v >> 16);
  p[2] = (char)(v >> 8);
  p[3] = (char)v;
}

inline uint32_t get_u32_be(const char* p) {
  return ((uint32_t)(uint8_t)p[0] << 24) | ((uint32_t)(uint8_t)p[1] << 16) |
         ((uint32_t)(uint8_t)p[2] << 8) | (uint32_t)(uint8_t)p[3];
}

}  // namespace wire
}  // namespace bam
