#include "base/codecs.h"

#include <string.h>

namespace bam {

// ---------------- base64 ----------------

static const char kB64[] = "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";

void Base64Encode(const std::string& in, std::string* out) {
  out->clear();
  out->reserve((in.size() + 2) / 3 * 4);
  size_t i = 0;
  while (i + 3 <= in.size()) {
    uint32_t v = ((uint8_t)in[i] << 16) | ((uint8_t)in[i + 1] << 8) | (uint8_t)in[i + 2];
    out->push_back(kB64[(v >> 18) & 63]);
    out->push_back(kB64[(v >> 12) & 63]);
    out->push_back(kB64[(v >> 6) & 63]);
    out->push_back(kB64[v & 63]);
    i += 3;
  }
  size_t rem = in.size() - i;
  if (rem == 1) {
    uint32_t v = (uint8_t)in[i] << 16;
    out->push_back(kB64[(v >> 18) & 63]);
    out->push_back(kB64[(v >> 12) & 63]);
    out->append("==");
  } else if (rem == 2) {
    uint32_t v = ((uint8_t)in[i] << 16) | ((uint8_t)in[i + 1] << 8);
    out->push_back(kB64[(v >> 18) & 63]);
    out->push_back(kB64[(v >> 12) & 63]);
    out->push_back(kB64[(v >> 6) & 63]);
    out->push_back('=');
  }
}

bool Base64Decode(const std::string& in, std::string* out) {
  static int8_t rev[256];
  static bool init = [] {
    memset(rev, -1, sizeof(rev));
    for (int i = 0; i < 64; ++i) rev[(uint8_t)kB64[i]] = (int8_t)i;
    return true;
  }();
  (void)init;
  out->clear();
  uint32_t acc = 0;
  int nbits = 0;
  for (char c : in) {
    if (c == '=' || c == '\n' || c == '\r') continue;
    int8_t v = rev[(uint8_t)c];
    if (v < 0) return false;
    acc = (acc << 6) | (uint32_t)v;
    nbits += 6;
    if (nbits >= 8) {
      nbits -= 8;
      out->push_back((char)(acc >> nbits));
    }
  }
  return true;
}

// ---------------- SHA-1 (FIPS 180-1, clean-room) ----------------

namespace {
inline uint32_t rol(uint32_t v, int n) { return (v << n) | (v >> (32 - n)); }
}  // namespace

std::string SHA1Hash(const std::string& input) {
  uint32_t h[5] = {0x67452301, 0xEFCDAB89, 0x98BADCFE, 0x10325476, 0xC3D2E1F0};
  uint64_t total_bits = (uint64_t)input.size() * 8;
  std::string msg = input;
  msg.push_back((char)0x80);
  while (msg.size() % 64 != 56) msg.push_back(0);
  for (int i = 7; i >= 0; --i) msg.push_back((char)(total_bits >> (i * 8)));
  for (size_t off = 0; off < msg.size(); off += 64) {
    uint32_t w[80];
    for (int i = 0; i < 16; ++i) {
      w[i] = ((uint8_t)msg[off + i * 4] << 24) | ((uint8_t)msg[off + i * 4 + 1] << 16) |
             ((uint8_t)msg[off + i * 4 + 2] << 8) | (uint8_t)msg[off + i * 4 + 3];
    }
    for (int i = 16; i < 80; ++i) w[i] = rol(w[i - 3] ^ w[i - 8] ^ w[i - 14] ^ w[i - 16], 1);
    uint32_t a = h[0], b = h[1], c = h[2], d = h[3], e = h[4];
    for (int i = 0; i < 80; ++i) {
      uint32_t f, k;
      if (i < 20) {
        f = (b & c) | ((~b) & d);
        k = 0x5A827999;
      } else if (i < 40) {
        f = b ^ c ^ d;
        k = 0x6ED9EBA1;
      } else if (i < 60) {
        f = (b & c) | (b & d) | (c & d);
        k = 0x8F1BBCDC;
      } else {
        f = b ^ c ^ d;
        k = 0xCA62C1D6;
      }
      uint32_t tmp = rol(a, 5) + f + e + k + w[i];
      e = d;
      d = c;
      c = rol(b, 30);
      b = a;
      a = tmp;
    }
    h[0] += a;
    h[1] += b;
    h[2] += c;
    h[3] += d;
    h[4] += e;
  }
  std::string digest(20, 0);
  for (int i = 0; i < 5; ++i) {
    digest[i * 4] = (char)(h[i] >> 24);
    digest[i * 4 + 1] = (char)(h[i] >> 16);
    digest[i * 4 + 2] = (char)(h[i] >> 8);
    digest[i * 4 + 3] = (char)h[i];
  }
  return digest;
}

std::string SHA1HexDigest(const std::string& input) {
  std::string d = SHA1Hash(input);
  std::string hex;
  static const char* k = "0123456789abcdef";
  for (unsigned char c : d) {
    hex.push_back(k[c >> 4]);
    hex.push_back(k[c & 15]);
  }
  return hex;
}

// ---------------- murmur3 ----------------

uint32_t MurmurHash3_32(const void* key, size_t len, uint32_t seed) {
  const uint8_t* data = (const uint8_t*)key;
  uint32_t h = seed;
  const uint32_t c1 = 0xcc9e2d51, c2 = 0x1b873593;
  size_t nblocks = len / 4;
  for (size_t i = 0; i < nblocks; ++i) {
    uint32_t k;
    memcpy(&k, data + i * 4, 4);
    k *= c1;
    k = rol(k, 15);
    k *= c2;
    h ^= k;
    h = rol(h, 13);
    h = h * 5 + 0xe6546b64;
  }
  uint32_t k = 0;
  const uint8_t* tail = data + nblocks * 4;
  switch (len & 3) {
    case 3:
      k ^= (uint32_t)tail[2] << 16;
      [[fallthrough]];
    case 2:
      k ^= (uint32_t)tail[1] << 8;
      [[fallthrough]];
    case 1:
      k ^= tail[0];
      k *= c1;
      k = rol(k, 15);
      k *= c2;
      h ^= k;
  }
  h ^= (uint32_t)len;
  h ^= h >> 16;
  h *= 0x85ebca6b;
  h ^= h >> 13;
  h *= 0xc2b2ae35;
  h ^= h >> 16;
  return h;
}

}  // namespace bam
