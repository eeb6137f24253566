// CRC32-C: SSE4.2 hardware path + slice-by-8 table fallback + GF(2) combine.
// Clean-room; algorithms are textbook (Castagnoli poly, reflected 0x82F63B78).
#include "base/crc32c.h"

#ifdef __SSE4_2__
#include <nmmintrin.h>
#endif

namespace bam {
namespace crc32c {

static const uint32_t kPoly = 0x82F63B78u;  // reflected Castagnoli

namespace {

struct Tables {
  uint32_t t[8][256];
  Tables() {
    for (uint32_t i = 0; i < 256; ++i) {
      uint32_t c = i;
      for (int k = 0; k < 8; ++k) c = (c >> 1) ^ ((c & 1) ? kPoly : 0);
      t[0][i] = c;
    }
    for (uint32_t i = 0; i < 256; ++i) {
      uint32_t c = t[0][i];
      for (int j = 1; j < 8; ++j) {
        c = t[0][c & 0xff] ^ (c >> 8);
        t[j][i] = c;
      }
    }
  }
};

const Tables& tables() {
  static Tables tb;
  return tb;
}

inline uint32_t crc_sw(uint32_t crc, const char* p, size_t n) {
  const Tables& tb = tables();
  const uint8_t* u = (const uint8_t*)p;
  // Process 8 bytes per step (slice-by-8).
  while (n >= 8) {
    uint64_t v;
    __builtin_memcpy(&v, u, 8);
    v ^= crc;  // low 4 bytes
    crc = tb.t[7][v & 0xff] ^ tb.t[6][(v >> 8) & 0xff] ^ tb.t[5][(v >> 16) & 0xff] ^
          tb.t[4][(v >> 24) & 0xff] ^ tb.t[3][(v >> 32) & 0xff] ^ tb.t[2][(v >> 40) & 0xff] ^
          tb.t[1][(v >> 48) & 0xff] ^ tb.t[0][(v >> 56) & 0xff];
    u += 8;
    n -= 8;
  }
  while (n--) crc = tb.t[0][(crc ^ *u++) & 0xff] ^ (crc >> 8);
  return crc;
}

}  // namespace

uint32_t Extend(uint32_t crc, const char* data, size_t n) {
  crc = ~crc;
#ifdef __SSE4_2__
  const uint8_t* p = (const uint8_t*)data;
  while (n && ((uintptr_t)p & 7)) {
    crc = _mm_crc32_u8(crc, *p++);
    --n;
  }
  while (n >= 8) {
    crc = (uint32_t)_mm_crc32_u64(crc, *(const uint64_t*)p);
    p += 8;
    n -= 8;
  }
  while (n--) crc = _mm_crc32_u8(crc, *p++);
  return ~crc;
#else
  return ~crc_sw(crc, data, n);
#endif
}

bool IsFastCrc32Supported() {
#ifdef __SSE4_2__
  return true;
#else
  return false;
#endif
}

// ---- Combine: crc(A|B) = crc_a * x^(8*len_b) + crc_b over GF(2) ----
// Standard matrix-exponentiation technique (as in zlib's crc32_combine).
namespace {

// Multiply the GF(2) 32x32 matrix `mat` by vector `vec`.
inline uint32_t gf2_matrix_times(const uint32_t* mat, uint32_t vec) {
  uint32_t sum = 0;
  while (vec) {
    if (vec & 1) sum ^= *mat;
    vec >>= 1;
    ++mat;
  }
  return sum;
}

inline void gf2_matrix_square(uint32_t* square, const uint32_t* mat) {
  for (int n = 0; n < 32; ++n) square[n] = gf2_matrix_times(mat, mat[n]);
}

}  // namespace

uint32_t Combine(uint32_t crc_a, uint32_t crc_b, size_t len_b) {
  if (len_b == 0) return crc_a;
  uint32_t even[32];  // x^(2n) operator
  uint32_t odd[32];   // x^n operator
  // odd = shift-by-one-bit operator (multiply by x).
  odd[0] = kPoly;
  uint32_t row = 1;
  for (int n = 1; n < 32; ++n) {
    odd[n] = row;
    row <<= 1;
  }
  gf2_matrix_square(even, odd);  // x^2
  gf2_matrix_square(odd, even);  // x^4
  // Apply len_b zero bytes (len_b*8 zero bits) to crc_a.
  uint32_t crc = crc_a;
  size_t len = len_b;
  do {
    gf2_matrix_square(even, odd);  // even = odd^2 (next power of two)
    if (len & 1) crc = gf2_matrix_times(even, crc);
    len >>= 1;
    if (len == 0) break;
    gf2_matrix_square(odd, even);
    if (len & 1) crc = gf2_matrix_times(odd, crc);
    len >>= 1;
  } while (len != 0);
  return crc ^ crc_b;
}

}  // namespace crc32c
}  // namespace bam
