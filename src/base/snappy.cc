// Clean-room snappy codec. Format (public spec, format_description.txt):
//   preamble: varint32 uncompressed length
//   elements: tag byte, low 2 bits = type
//     00 literal: len-1 in tag>>2 if <60; else 60..63 -> 1..4 LE extra bytes
//     01 copy, 1-byte offset: len 4..11 = ((tag>>2)&7)+4, off 11 bits =
//        ((tag>>5)<<8) | next byte
//     10 copy, 2-byte offset: len = (tag>>2)+1 (1..64), off = 2 LE bytes
//     11 copy, 4-byte offset: len = (tag>>2)+1, off = 4 LE bytes
// Compressor: greedy per-64KiB block, 4-byte hash chains with the standard
// skip acceleration; emits only 1/2-byte-offset copies.
#include "base/snappy.h"

#include <string.h>

#include <vector>

namespace bam {
namespace snappy {

namespace {

const size_t kBlockSize = 64 * 1024;
const int kHashBits = 14;
const size_t kHashTableSize = (size_t)1 << kHashBits;

inline uint32_t load32(const char* p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;
}

inline uint64_t load64(const char* p) {
  uint64_t v;
  memcpy(&v, p, 8);
  return v;
}

inline uint32_t hash4(uint32_t v) { return (v * 0x1e35a7bd) >> (32 - kHashBits); }

inline char* emit_varint(char* dst, uint64_t v) {
  while (v >= 0x80) {
    *dst++ = (char)(v | 0x80);
    v >>= 7;
  }
  *dst++ = (char)v;
  return dst;
}

char* emit_literal(char* dst, const char* src, size_t len) {
  if (len == 0) return dst;
  size_t n = len - 1;
  if (n < 60) {
    *dst++ = (char)(n << 2);
  } else {
    int count = 0;
    size_t tmp = n;
    char bytes[4];
    while (tmp > 0 || count == 0) {
      bytes[count++] = (char)(tmp & 0xff);
      tmp >>= 8;
      if (count == 4) break;
      if (tmp == 0) break;
    }
    *dst++ = (char)((59 + count) << 2);
    for (int i = 0; i < count; ++i) *dst++ = bytes[i];
  }
  memcpy(dst, src, len);
  return dst + len;
}

char* emit_copy_upto64(char* dst, size_t offset, size_t len) {
  // len in [4, 64] here (callers split longer)
  if (len < 12 && offset < 2048) {
    *dst++ = (char)(1 | ((len - 4) << 2) | ((offset >> 8) << 5));
    *dst++ = (char)(offset & 0xff);
  } else {
    *dst++ = (char)(2 | ((len - 1) << 2));
    *dst++ = (char)(offset & 0xff);
    *dst++ = (char)((offset >> 8) & 0xff);
  }
  return dst;
}

char* emit_copy(char* dst, size_t offset, size_t len) {
  while (len >= 68) {
    dst = emit_copy_upto64(dst, offset, 64);
    len -= 64;
  }
  if (len > 64) {
    dst = emit_copy_upto64(dst, offset, 60);
    len -= 60;
  }
  return emit_copy_upto64(dst, offset, len);
}

}  // namespace

size_t MaxCompressedLength(size_t n) { return 32 + n + n / 6; }

size_t RawCompress(const char* src, size_t n, char* dst) {
  char* out = emit_varint(dst, n);
  std::vector<uint16_t> table(kHashTableSize);
  size_t pos = 0;
  while (pos < n) {
    size_t block_end = pos + kBlockSize < n ? pos + kBlockSize : n;
    const char* base = src + pos;
    size_t block_len = block_end - pos;
    memset(table.data(), 0, kHashTableSize * sizeof(uint16_t));
    size_t ip = 0;                 // cursor within block
    size_t next_emit = 0;          // first unemitted literal byte
    if (block_len >= 15) {
      size_t skip = 32;            // skip acceleration (1 byte per 32 misses)
      size_t candidate = 0;
      for (ip = 1; ip + 4 <= block_len - 4;) {
        uint32_t cur = load32(base + ip);
        uint32_t h = hash4(cur);
        candidate = table[h];
        table[h] = (uint16_t)ip;
        if (candidate != 0 && load32(base + candidate) == cur && candidate < ip) {
          // match: emit pending literals then extend
          out = emit_literal(out, base + next_emit, ip - next_emit);
          size_t mlen = 4;
          while (ip + mlen < block_len && base[candidate + mlen] == base[ip + mlen]) ++mlen;
          out = emit_copy(out, ip - candidate, mlen);
          ip += mlen;
          next_emit = ip;
          skip = 32;
          continue;
        }
        ip += 1 + (skip >> 5);
        skip += 1;  // slowly accelerate through incompressible data
      }
    }
    // trailing literals of the block
    out = emit_literal(out, base + next_emit, block_len - next_emit);
    pos = block_end;
  }
  return out - dst;
}

bool GetUncompressedLength(const char* p, size_t n, size_t* result) {
  uint64_t v = 0;
  int shift = 0;
  size_t i = 0;
  while (i < n && shift < 35) {
    uint8_t b = (uint8_t)p[i++];
    v |= (uint64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) {
      *result = (size_t)v;
      return true;
    }
    shift += 7;
  }
  return false;
}

bool RawUncompress(const char* p, size_t n, char* dst) {
  // skip preamble
  size_t i = 0;
  while (i < n && (p[i] & 0x80)) ++i;
  if (i >= n) return false;
  ++i;
  size_t expected;
  if (!GetUncompressedLength(p, n, &expected)) return false;
  size_t op = 0;
  while (i < n) {
    uint8_t tag = (uint8_t)p[i++];
    int type = tag & 3;
    if (type == 0) {  // literal
      size_t len = (tag >> 2) + 1;
      if (len > 60) {
        int extra = (int)len - 60;
        if (i + extra > n) return false;
        len = 0;
        for (int k = 0; k < extra; ++k) len |= (size_t)(uint8_t)p[i + k] << (8 * k);
        len += 1;
        i += extra;
      }
      if (i + len > n || op + len > expected) return false;
      memcpy(dst + op, p + i, len);
      i += len;
      op += len;
    } else {
      size_t len, offset;
      if (type == 1) {
        if (i >= n) return false;
        len = ((tag >> 2) & 7) + 4;
        offset = ((size_t)(tag >> 5) << 8) | (uint8_t)p[i++];
      } else if (type == 2) {
        if (i + 2 > n) return false;
        len = (tag >> 2) + 1;
        offset = (uint8_t)p[i] | ((size_t)(uint8_t)p[i + 1] << 8);
        i += 2;
      } else {
        if (i + 4 > n) return false;
        len = (tag >> 2) + 1;
        offset = (uint8_t)p[i] | ((size_t)(uint8_t)p[i + 1] << 8) |
                 ((size_t)(uint8_t)p[i + 2] << 16) | ((size_t)(uint8_t)p[i + 3] << 24);
        i += 4;
      }
      if (offset == 0 || offset > op || op + len > expected) return false;
      // overlapping copies must proceed byte-wise
      const char* from = dst + op - offset;
      char* to = dst + op;
      if (offset >= len) {
        memcpy(to, from, len);
      } else {
        for (size_t k = 0; k < len; ++k) to[k] = from[k];
      }
      op += len;
    }
  }
  return op == expected;
}

void Compress(const char* src, size_t n, std::string* out) {
  out->resize(MaxCompressedLength(n));
  size_t sz = RawCompress(src, n, &(*out)[0]);
  out->resize(sz);
}

bool Uncompress(const char* compressed, size_t n, std::string* out) {
  size_t len;
  if (!GetUncompressedLength(compressed, n, &len)) return false;
  // Malloc-bomb guard (found by tests/fuzz/fuzz_snappy.cc): the preamble
  // can claim any length; real snappy streams expand at most 64x per
  // 2-byte copy tag, so a claim beyond 64*n is certainly corrupt — reject
  // BEFORE allocating.
  if (len > 64 * n + 1024) return false;
  out->resize(len);
  return RawUncompress(compressed, n, len > 0 ? &(*out)[0] : (char*)"");
}

}  // namespace snappy
}  // namespace bam
