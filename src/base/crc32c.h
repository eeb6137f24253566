// brpc_amd: CRC32-C (Castagnoli) — host reference path.
// Parity: reference butil/crc32c.h (Value/Extend). Adds Combine() — the
// GF(2) concatenation operator used to merge per-chunk CRCs computed in
// parallel by the gfx950 kernel (hip/crc32c.hip).
#pragma once

#include <stddef.h>
#include <stdint.h>

namespace bam {
namespace crc32c {

// CRC of data[0,n) given the crc of a preceding byte stream.
uint32_t Extend(uint32_t init_crc, const char* data, size_t n);

inline uint32_t Value(const char* data, size_t n) { return Extend(0, data, n); }

// CRC(A|B) from crc_a = CRC(A), crc_b = CRC(B), len_b = |B|.
uint32_t Combine(uint32_t crc_a, uint32_t crc_b, size_t len_b);

// True if the SSE4.2 hardware path is compiled in and used.
bool IsFastCrc32Supported();

}  // namespace crc32c
}  // namespace bam
