// Philox4x32-10 counter-based RNG (device-side, stateless per element).
// Used by the dropout kernels: mask bits are regenerated in backward from
// the same (seed, offset) pair, so no mask tensor is ever stored
// (SURVEY.md K5). Offsets come from a device-side counter so the whole
// step stays graph-capturable.
#pragma once

#include <cstdint>

#include "common.h"

namespace zamd {

DEV_INLINE uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hi) {
  uint64_t p = (uint64_t)a * b;
  *hi = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

struct Philox4 {
  uint32_t x, y, z, w;
};

DEV_INLINE Philox4 philox4x32_10(uint64_t seed, uint64_t counter) {
  constexpr uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  constexpr uint32_t B0 = 0x9E3779B9u, B1 = 0xBB67AE85u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  uint32_t c0 = (uint32_t)counter, c1 = (uint32_t)(counter >> 32);
  uint32_t c2 = 0, c3 = 0;
#pragma unroll
  for (int round = 0; round < 10; ++round) {
    uint32_t hi0, hi1;
    uint32_t lo0 = mulhilo(M0, c0, &hi0);
    uint32_t lo1 = mulhilo(M1, c2, &hi1);
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += B0; k1 += B1;
  }
  return {c0, c1, c2, c3};
}

// uniform in [0, 1)
DEV_INLINE float u32_to_uniform(uint32_t v) {
  return (float)(v >> 8) * (1.0f / 16777216.0f);
}

}  // namespace zamd
