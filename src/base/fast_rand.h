// brpc_amd: thread-local xorshift128+ PRNG (parity: reference butil/fast_rand.h).
#pragma once

#include <stdint.h>
#include <time.h>
#include <unistd.h>

namespace bam {

struct FastRandState {
  uint64_t s0, s1;
};

inline uint64_t splitmix64(uint64_t& x) {
  x += 0x9E3779B97f4A7C15ULL;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

inline uint64_t fast_rand(FastRandState& st) {
  uint64_t x = st.s0;
  const uint64_t y = st.s1;
  st.s0 = y;
  x ^= x << 23;
  st.s1 = x ^ y ^ (x >> 17) ^ (y >> 26);
  return st.s1 + y;
}

inline FastRandState& tls_rand_state() {
  static thread_local FastRandState st = [] {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    uint64_t seed = (uint64_t)ts.tv_nsec ^ ((uint64_t)getpid() << 32) ^ (uint64_t)(uintptr_t)&ts;
    FastRandState s;
    s.s0 = splitmix64(seed);
    s.s1 = splitmix64(seed);
    return s;
  }();
  return st;
}

inline uint64_t fast_rand() { return fast_rand(tls_rand_state()); }

// Uniform in [0, range). range == 0 returns 0.
inline uint64_t fast_rand_less_than(uint64_t range) {
  if (range == 0) return 0;
  return fast_rand() % range;  // modulo bias negligible for LB use
}

inline double fast_rand_double() {
  return (fast_rand() >> 11) * (1.0 / 9007199254740992.0);  // 53-bit mantissa
}

}  // namespace bam
