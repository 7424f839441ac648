// Action-sampling kernels for the rollout hot loop.
//
// The reference samples actions through torch.distributions on the
// host path (stochastic_policy.py:26-41): exp(log_std) + randn + mul +
// add (+ clamp) is 4-5 kernels per sampler step.  Here the sample is
// ONE kernel: counter-based Philox4x32-10 RNG (stateless, seeded from
// the run seed + a per-call offset -> bitwise reproducible, replay-safe
// under hipGraph capture) with Box-Muller for normals.
#include "common.h"
#include "philox.h"

// actions = mean + exp(log_std) * z, with optional clip to [-limit, limit]
// (limit < 0 disables clipping; noise_scale overrides sigma when >= 0,
// which implements the _NoisedPolicy deterministic-actor case)
__global__ __launch_bounds__(256) void gaussian_sample_kernel(
    const float* __restrict__ mean, const float* __restrict__ log_std,
    float* __restrict__ out, int B, int D, uint64_t seed, uint64_t offset,
    float noise_scale, float limit,
    const unsigned long long* __restrict__ offset_ptr) {
  // offset_ptr: optional device counter added to `offset` -- lets a
  // hipGraph replay draw fresh randomness (the by-value offset is
  // frozen at capture; the counter advances via counter_add_kernel)
  if (offset_ptr) offset += *offset_ptr;
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
