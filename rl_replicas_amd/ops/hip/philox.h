// Counter-based Philox4x32-10 RNG + Box-Muller (Salmon et al. 2011).
// Stateless: keyed by (seed, offset, index) -> bitwise-reproducible and
// replay-safe under hipGraph capture.  Shared by the sampling and
// device-env kernels.
#pragma once
#include "common.h"

DEV_INLINE void philox_round(uint32_t* c, uint32_t* k) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  uint32_t hi0 = __umulhi(M0, c[0]), lo0 = M0 * c[0];
  uint32_t hi1 = __umulhi(M1, c[2]), lo1 = M1 * c[2];
  uint32_t n0 = hi1 ^ c[1] ^ k[0];
  uint32_t n1 = lo1;
  uint32_t n2 = hi0 ^ c[3] ^ k[1];
  uint32_t n3 = lo0;
  c[0] = n0; c[1] = n1; c[2] = n2; c[3] = n3;
}

DEV_INLINE void philox4(uint64_t seed, uint64_t offset, uint32_t idx, uint32_t* out) {
  uint32_t c[4] = {idx, (uint32_t)offset, (uint32_t)(offset >> 32), 0u};
  uint32_t k[2] = {(uint32_t)seed, (uint32_t)(seed >> 32)};
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
  #pragma unroll
  for (int r = 0; r < 10; ++r) {
    philox_round(c, k);
    k[0] += W0; k[1] += W1;
  }
  out[0] = c[0]; out[1] = c[1]; out[2] = c[2]; out[3] = c[3];
}

DEV_INLINE float u32_to_open_unit(uint32_t x) {
  // (0, 1]: avoids log(0) in Box-Muller
  return ((float)x + 1.0f) * (1.0f / 4294967296.0f);
}

DEV_INLINE void box_muller(uint32_t a, uint32_t b, float* z0, float* z1) {
  const float u1 = u32_to_open_unit(a);
  const float u2 = u32_to_open_unit(b);
  const float r = sqrtf(-2.f * __logf(u1));
  float s, c;
  __sincosf(6.2831853071795864f * u2, &s, &c);
  *z0 = r * c;
  *z1 = r * s;
}

// one standard normal keyed by (seed, offset, idx)
DEV_INLINE float philox_normal(uint64_t seed, uint64_t offset, uint32_t idx) {
  uint32_t r[4];
  philox4(seed, offset, idx, r);
  float z0, z1;
  box_muller(r[0], r[1], &z0, &z1);
  return z0;
}
