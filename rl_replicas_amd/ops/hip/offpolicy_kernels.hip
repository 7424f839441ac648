// Off-policy (DDPG/TD3) hot-loop kernels.
//
// The reference's 50-iteration train loop (ddpg.py:195-253,
// td3.py:214-263) does per iteration: numpy minibatch gather + 5x
// from_numpy, a target chain (randn + clamp + add + clamp + two target-Q
// forwards + min + r+gamma(1-d)q), two critic MSE steps and a delayed
// actor step.  On MI355X the whole loop is device-resident; these
// kernels cover the three pieces the fused-MLP/loss/Adam kernels don't:
//
//   replay_gather_kernel  one kernel draws the minibatch indices
//                         (counter-based Philox -> graph-replay safe)
//                         and gathers obs|act (pre-concatenated as the
//                         critic input), next_obs, rewards, dones from
//                         the HBM ring -- replaces randint + 5
//                         index_selects + torch.cat
//   td3_smooth_kernel     target-policy smoothing: clamp(a +
//                         clamp(scale*z, +-clip), +-limit) in one pass
//   q_target_min2_kernel  min-twin bootstrap target
//
// All three read their RNG offset through an optional device counter so
// a hipGraph replay of the whole train loop draws fresh randomness
// (same pattern as sample_kernels.hip).
#include "common.h"
#include "philox.h"

// out = clamp(a + clamp(scale * z, -clip, clip), -limit, limit)
// (reference td3.py:325-341)
__global__ __launch_bounds__(256) void td3_smooth_kernel(
    const float* __restrict__ a, float* __restrict__ out, int total,
    uint64_t seed, uint64_t offset, float scale, float clip, float limit,
    const unsigned long long* __restrict__ offset_ptr) {
  if (offset_ptr) offset += *offset_ptr;
  for (int i = blockIdx.x * 256 + threadIdx.x; i < total;
       i += gridDim.x * 256) {
    float eps = scale * philox_normal(seed, offset, (uint32_t)i);
    eps = fminf(fmaxf(eps, -clip), clip);
    float v = a[i] + eps;
    out[i] = fminf(fmaxf(v, -limit), limit);
  }
}

// out = r + gamma * (1 - d) * min(q1, q2)   (reference td3.py:335-341)
__global__ __launch_bounds__(256) void q_target_min2_kernel(
    const float* __restrict__ r, const float* __restrict__ d,
    const float* __restrict__ q1, const float* __restrict__ q2,
    float* __restrict__ out, float gamma, int n) {
  for (int i = blockIdx.x * 256 + threadIdx.x; i < n; i += gridDim.x * 256) {
    out[i] = r[i] + gamma * (1.f - d[i]) * fminf(q1[i], q2[i]);
  }
}

// Minibatch gather from the HBM replay ring (replay_buffer.py).
// Row i's index is Philox-keyed by (seed, offset, i) modulo the CURRENT
// buffer size, which is read from a device scalar so one captured graph
// keeps sampling correctly as the ring fills between epochs.  Each row
// writes:
//   qin[i]      = [obs(idx) | act(idx)]   (critic input, pre-concatenated)
//   obs_out[i]  = obs(idx)                (actor input, contiguous)
//   nxt_out[i]  = next_obs(idx)
//   rew_out[i], dn_out[i]
__global__ __launch_bounds__(256) void replay_gather_kernel(
    ReplayGatherArgs a) {
  uint64_t offset = a.offset;
  if (a.offset_ptr) offset += *a.offset_ptr;
  const uint32_t size = (uint32_t)*a.size;
  const int row_elems = 2 * a.O + a.A + 2;
  const long total = (long)a.B * row_elems;
  for (long t = blockIdx.x * 256 + threadIdx.x; t < total;
       t += (long)gridDim.x * 256) {
    const int row = (int)(t / row_elems);
    const int e = (int)(t % row_elems);
    uint32_t r4[4];
    philox4(a.seed, offset, (uint32_t)row, r4);
    const long idx = (long)(r4[0] % size);
    if (e < a.O) {
      const float v = a.obs[idx * a.O + e];
      a.qin[(long)row * (a.O + a.A) + e] = v;
      a.obs_out[(long)row * a.O + e] = v;
    } else if (e < a.O + a.A) {
      const int c = e - a.O;
      a.qin[(long)row * (a.O + a.A) + a.O + c] = a.act[idx * a.A + c];
    } else if (e < 2 * a.O + a.A) {
      const int c = e - a.O - a.A;
      a.nxt_out[(long)row * a.O + c] = a.nxt[idx * a.O + c];
    } else if (e == 2 * a.O + a.A) {
      a.rew_out[row] = a.rew[idx];
    } else {
      a.dn_out[row] = a.dn[idx];
    }
  }
}
