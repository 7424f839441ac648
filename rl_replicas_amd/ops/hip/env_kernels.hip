// GPU-resident synthetic-environment kernels (envs/device.py fast path).
//
// One rollout step of the synthetic MuJoCo-shaped envs
// (synthetic.py:72-78) is
//     s' = tanh(s A + clip(a) B + sigma * eps)
//     r  = <s', w> - 0.1 |clip(a)|^2
// which the eager torch path issues as ~10 tiny kernels.  Here it is
// ONE launch: one wavefront per env instance computes the row's GEMV
// columns, the tanh epilogue, the Philox dynamics noise, the reward
// reduction, and (on lockstep horizon boundaries) the autoreset init
// state — N x O is a few thousand elements, so this regime is pure
// launch-latency: fewer launches IS the optimization.
#include "common.h"
#include "philox.h"

// grid = N blocks of 64 threads (one wave per env row).
// s_out: next live state (post-autoreset when do_reset).
// final_out: true successor state (pre-reset; GAE bootstrap).  May
// alias s_out when do_reset == 0.
__global__ __launch_bounds__(64) void synthetic_env_step_kernel(
    const float* __restrict__ state, const float* __restrict__ actions,
    const float* __restrict__ A, const float* __restrict__ Bm,
    const float* __restrict__ w, float* __restrict__ s_out,
    float* __restrict__ final_out, float* __restrict__ reward, int N, int O,
    int Adim, float sigma, uint64_t seed, uint64_t offset, int do_reset,
    const unsigned long long* __restrict__ offset_ptr) {
  if (offset_ptr) offset += *offset_ptr;  // hipGraph-replay RNG advance
  const int row = blockIdx.x;
  if (row >= N) return;
  const float* s = state + (long)row * O;
  const float* a = actions + (long)row * Adim;
  const int o = threadIdx.x;

  float r_part = 0.f;
  float out_o = 0.f;
  if (o < O) {
    float acc = 0.f;
    for (int k = 0; k < O; ++k) acc += s[k] * A[k * O + o];
    for (int j = 0; j < Adim; ++j) {
      const float aj = fminf(fmaxf(a[j], -1.f), 1.f);
      acc += aj * Bm[j * O + o];
    }
    if (sigma > 0.f)
      acc += sigma * philox_normal(seed, offset, (uint32_t)(row * O + o));
    out_o = tanhf(acc);
    final_out[(long)row * O + o] = out_o;
    r_part = out_o * w[o];
  }
  if (o == 0) {
    float pen = 0.f;
    for (int j = 0; j < Adim; ++j) {
      const float aj = fminf(fmaxf(a[j], -1.f), 1.f);
      pen += aj * aj;
    }
    r_part -= 0.1f * pen;
  }
  const float rsum = wave_reduce_sum(r_part);
  if (o == 0) reward[row] = rsum;
  if (o < O) {
    // offset+1 is the autoreset noise stream (host bumps the counter by 2
    // on reset steps so streams never collide)
    s_out[(long)row * O + o] =
        do_reset ? 0.1f * philox_normal(seed, offset + 1, (uint32_t)(row * O + o))
                 : out_o;
  }
}

// fresh init states: s = 0.1 * eps  (SyntheticEnv._init_state)
__global__ __launch_bounds__(256) void synthetic_env_reset_kernel(
    float* __restrict__ s_out, int total, uint64_t seed, uint64_t offset,
    const unsigned long long* __restrict__ offset_ptr) {
  if (offset_ptr) offset += *offset_ptr;
  for (int i = blockIdx.x * 256 + threadIdx.x; i < total; i += gridDim.x * 256)
    s_out[i] = 0.1f * philox_normal(seed, offset, (uint32_t)i);
}

// advance a device RNG counter (graph-replayable: one bump per epoch)
__global__ void counter_add_kernel(unsigned long long* ctr,
                                   unsigned long long delta) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *ctr += delta;
}
