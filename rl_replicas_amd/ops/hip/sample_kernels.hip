// Action-sampling kernels for the rollout hot loop.
//
// The reference samples actions through torch.distributions on the
// host path (stochastic_policy.py:26-41): exp(log_std) + randn + mul +
// add (+ clamp) is 4-5 kernels per sampler step.  Here the sample is
// ONE kernel: counter-based Philox4x32-10 RNG (stateless, seeded from
// the run seed + a per-call offset -> bitwise reproducible, replay-safe
// under hipGraph capture) with Box-Muller for normals.
#include "common.h"

// ---- Philox4x32-10 (Salmon et al. 2011 constants) ----
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

// actions = mean + exp(log_std) * z, with optional clip to [-limit, limit]
// (limit < 0 disables clipping; noise_scale overrides sigma when >= 0,
// which implements the _NoisedPolicy deterministic-actor case)
__global__ __launch_bounds__(256) void gaussian_sample_kernel(
    const float* __restrict__ mean, const float* __restrict__ log_std,
    float* __restrict__ out, int B, int D, uint64_t seed, uint64_t offset,
    float noise_scale, float limit) {
  const int total = B * D;
  for (int i = blockIdx.x * 256 + threadIdx.x; i < total; i += gridDim.x * 256) {
    uint32_t r[4];
    philox4(seed, offset, (uint32_t)i, r);
    float z0, z1;
    box_muller(r[0], r[1], &z0, &z1);
    const int d = i % D;
    const float sigma = noise_scale >= 0.f ? noise_scale : __expf(log_std[d]);
    float a = mean[i] + sigma * z0;
    if (limit >= 0.f) a = fminf(fmaxf(a, -limit), limit);
    out[i] = a;
  }
}

// categorical sample: inverse-CDF over softmax(logits) per row
__global__ __launch_bounds__(256) void categorical_sample_kernel(
    const float* __restrict__ logits, int64_t* __restrict__ out, int B, int N,
    uint64_t seed, uint64_t offset) {
  for (int row = blockIdx.x * 256 + threadIdx.x; row < B; row += gridDim.x * 256) {
    const float* lg = logits + (long)row * N;
    float m = lg[0];
    for (int j = 1; j < N; ++j) m = fmaxf(m, lg[j]);
    float z = 0.f;
    for (int j = 0; j < N; ++j) z += __expf(lg[j] - m);
    uint32_t r[4];
    philox4(seed, offset, (uint32_t)row, r);
    const float u = u32_to_open_unit(r[0]) * z;
    float acc = 0.f;
    int pick = N - 1;
    for (int j = 0; j < N; ++j) {
      acc += __expf(lg[j] - m);
      if (u <= acc) { pick = j; break; }
    }
    out[row] = pick;
  }
}
