// Rollout-processing kernels: segmented discounted scans (GAE + returns),
// advantage normalization, and the fused Q-target.
//
// MI355X replacement for the reference's per-episode host pipeline
// (scipy IIR scans in utils.py:14-44, bootstrap utils.py:74-87, torch
// normalize utils.py:90-92): the WHOLE ragged rollout is processed in
// one launch.  A first-order backward recurrence acc_t = v_t + c*acc_{t+1}
// is associative under (scale, offset) composition
//     (a1,b1) o (a2,b2) = (a1*a2, b1 + a1*b2),
// so each episode runs as an LDS-staged parallel scan: one workgroup per
// episode, each thread owns a contiguous chunk (backward local scan),
// thread-level carries are composed with a wave shuffle scan + LDS wave
// carries, then chunks are replayed with their incoming carry.  Both
// recurrences (returns with gamma over bootstrapped rewards; advantages
// with gamma*lambda over TD deltas) run in the same kernel pass.
#include "common.h"

struct ScanPair {
  float a;  // accumulated scale
  float b;  // accumulated offset
};

DEV_INLINE ScanPair compose(ScanPair lo, ScanPair hi) {
  // apply `hi` (later in time) then `lo`: acc = lo.b + lo.a*(hi.b + hi.a*acc)
  return {lo.a * hi.a, lo.b + lo.a * hi.b};
}

// Inclusive backward scan of (a,b) pairs across the workgroup's threads:
// thread t receives the composition of pairs of threads t..nthreads-1.
// Returns the EXCLUSIVE carry (composition of threads t+1..nthreads-1).
template <int NTHREADS>
DEV_INLINE ScanPair block_suffix_scan_exclusive(ScanPair mine, int tid) {
  __shared__ ScanPair wave_carry[NTHREADS / WAVE];
  const int lane = tid & 63;
  // wave-level inclusive suffix scan via shuffles (reverse order)
  ScanPair inc = mine;
  #pragma unroll
  for (int off = 1; off < WAVE; off <<= 1) {
    float oa = __shfl_down(inc.a, off, WAVE);
    float ob = __shfl_down(inc.b, off, WAVE);
    if (lane + off < WAVE) inc = compose(inc, {oa, ob});
  }
  // wave 'inc' at lane 0 = whole wave's composition
  const int wid = tid / WAVE;
  if (lane == 0) wave_carry[wid] = inc;
  __syncthreads();
  // serial suffix-compose the wave carries (few waves); smaller wave
  // index = later application = outermost
  ScanPair carry = {1.f, 0.f};  // identity
  #pragma unroll
  for (int w = NTHREADS / WAVE - 1; w >= 0; --w) {
    if (w > wid) carry = compose(wave_carry[w], carry);
  }
  // exclusive within the wave: composition of lanes (lane+1..63)
  ScanPair excl = {1.f, 0.f};
  float oa = __shfl_down(inc.a, 1, WAVE);
  float ob = __shfl_down(inc.b, 1, WAVE);
  if (lane < WAVE - 1) excl = {oa, ob};
  __syncthreads();
  return compose(excl, carry);
}

// One workgroup per episode.
#define SCAN_THREADS 256

__global__ __launch_bounds__(SCAN_THREADS) void segmented_gae_kernel(
    const float* __restrict__ rewards, const float* __restrict__ values,
    const float* __restrict__ last_values, const int* __restrict__ offsets,
    const int* __restrict__ dones, float* __restrict__ advantages,
    float* __restrict__ returns, float gamma, float lam) {
  const int e = blockIdx.x;
  const int lo = offsets[e];
  const int hi = offsets[e + 1];
  const int len = hi - lo;
  if (len <= 0) return;
  const int tid = threadIdx.x;
  const float last_v = last_values[e];
  const float boot = dones[e] ? 0.f : last_v;
  const float gl = gamma * lam;

  // contiguous chunk per thread
  const int chunk = (len + SCAN_THREADS - 1) / SCAN_THREADS;
  const int c0 = tid * chunk;
  const int c1 = min(c0 + chunk, len);

  // ---- pass A: returns (coef gamma over rewards, seed = boot) ----
  // local backward composition over my chunk
  ScanPair mine = {1.f, 0.f};
  for (int t = c0; t < c1; ++t) {  // forward order composes correctly:
    // earlier t must be OUTERMOST: acc_t = r_t + g*acc_{t+1}
    mine = (t == c0) ? ScanPair{gamma, rewards[lo + t]}
                     : compose(mine, {gamma, rewards[lo + t]});
  }
  ScanPair carry = block_suffix_scan_exclusive<SCAN_THREADS>(mine, tid);
  // incoming accumulator for my chunk's LAST element = carry applied to boot
  float acc = carry.b + carry.a * boot;
  for (int t = c1 - 1; t >= c0; --t) {
    acc = rewards[lo + t] + gamma * acc;
    returns[lo + t] = acc;
  }
  __syncthreads();

  // ---- pass B: advantages (coef gamma*lam over TD deltas, seed = 0) ----
  auto delta_at = [&](int t) {
    float v_next = (t + 1 < len) ? values[lo + t + 1] : last_v;
    return rewards[lo + t] + gamma * v_next - values[lo + t];
  };
  mine = {1.f, 0.f};
  for (int t = c0; t < c1; ++t) {
    mine = (t == c0) ? ScanPair{gl, delta_at(t)} : compose(mine, {gl, delta_at(t)});
  }
  carry = block_suffix_scan_exclusive<SCAN_THREADS>(mine, tid);
  acc = carry.b;  // seed 0
  for (int t = c1 - 1; t >= c0; --t) {
    acc = delta_at(t) + gl * acc;
    advantages[lo + t] = acc;
  }
}

// ---------------------------------------------------------------------------
// single-workgroup normalize: y = (x - mean) / std (Bessel), deterministic
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(1024) void normalize_kernel(
    const float* __restrict__ x, float* __restrict__ y, int n) {
  __shared__ float red[1024 / WAVE];
  __shared__ float s_mean, s_rstd;
  const int tid = threadIdx.x;

  float sum = 0.f, sumsq = 0.f;
  for (int i = tid; i < n; i += 1024) {
    float v = x[i];
    sum += v;
    sumsq += v * v;
  }
  // two reductions through LDS (deterministic order)
  for (int pass = 0; pass < 2; ++pass) {
    float v = pass == 0 ? sum : sumsq;
    v = wave_reduce_sum(v);
    if ((tid & 63) == 0) red[tid / WAVE] = v;
    __syncthreads();
    if (tid == 0) {
      float total = 0.f;
      for (int w = 0; w < 1024 / WAVE; ++w) total += red[w];
      if (pass == 0) s_mean = total / n;
      else {
        float var = (total - (float)n * s_mean * s_mean) / (n - 1);
        s_rstd = rsqrtf(var);
      }
    }
    __syncthreads();
  }
  const float mean = s_mean, rstd = s_rstd;
  for (int i = tid; i < n; i += 1024) y[i] = (x[i] - mean) * rstd;
}

// ---------------------------------------------------------------------------
// fused Q-learning target: y = r + gamma * (1 - done) * q_next
// ---------------------------------------------------------------------------
__global__ void q_target_kernel(const float* __restrict__ r,
                                const float* __restrict__ d,
                                const float* __restrict__ qn,
                                float* __restrict__ out, float gamma, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = r[i] + gamma * (1.f - d[i]) * qn[i];
}
