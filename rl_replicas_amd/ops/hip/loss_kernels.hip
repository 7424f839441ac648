// Fused policy/value loss forward+backward kernels.
//
// Replaces the ~15-kernel torch-eager chains of the reference's loss
// pipelines (PPO clipped surrogate ppo.py:237-257, VPG score loss
// vpg.py:200-203, value MSE ppo.py:283-287, approximate KL
// ppo.py:259-269) with one kernel per loss that computes the loss
// scalar AND the analytic input gradients (dmean / dlogits / dv,
// dlog_std) directly — the MLP backward kernels consume those, so the
// whole training step never touches torch autograd on the GPU path.
//
// Gradient semantics replicate torch EXACTLY, including the tie rule of
// torch.min (0.5/0.5 split — ties are the COMMON case inside the PPO
// clip region where ratio == clipped ratio) and torch.clamp's
// boundary-inclusive gradient (verified against torch 2.10).
//
// All kernels are single-workgroup (1024 threads) with fixed-order LDS
// reductions: bitwise deterministic, and at B<=32K rows the whole loss
// is latency-bound anyway.
#include "common.h"

#define LOSS_THREADS 1024

// deterministic block-sum over LOSS_THREADS threads; returns total on
// every thread (LDS wave partials summed in fixed order)
DEV_INLINE float block_sum(float v, float* red /*[LOSS_THREADS/WAVE]*/) {
  v = wave_reduce_sum(v);
  const int tid = threadIdx.x;
  if ((tid & 63) == 0) red[tid / WAVE] = v;
  __syncthreads();
  float total = 0.f;
  #pragma unroll
  for (int w = 0; w < LOSS_THREADS / WAVE; ++w) total += red[w];
  __syncthreads();
  return total;
}

// ---------------------------------------------------------------------------
// Gaussian policy (state-independent log_std — gaussian_policy.py:18-35):
// logp(a|s) = sum_d [ -0.5 z_d^2 - log_std_d ] - D/2 log(2pi),  z=(a-mean)/sigma
//
// Multi-block phase 1 (templated compile-time D -> register-resident
// dlog_std accumulators) writing per-block partials; a tiny finalize
// kernel sums the <=LOSS_BLOCKS partials in fixed order (deterministic).
// outputs: dmean[B,D]; partials[blk][D+1] = (dlog_std_d..., loss)
// ---------------------------------------------------------------------------
#define LOSS_BLOCKS 32
#define LOSS_P_THREADS 256
#define GAUSS_MAX_D 512  // generic-D kernel's LDS sigma table bound

template <int DT>
__global__ __launch_bounds__(LOSS_P_THREADS) void gaussian_policy_loss_bwd_t(
    const float* __restrict__ mean, const float* __restrict__ actions,
    const float* __restrict__ old_logp, const float* __restrict__ adv,
    const float* __restrict__ log_std, float* __restrict__ dmean,
    float* __restrict__ partials, int B, float clip, int mode) {
  __shared__ float red[LOSS_P_THREADS / WAVE];
  const int tid = threadIdx.x;
  const int gid = blockIdx.x * LOSS_P_THREADS + tid;

  float sigma[DT], inv_s2[DT];
  float base = 0.5f * (float)DT * LOG_2PI;
  #pragma unroll
  for (int d = 0; d < DT; ++d) {
    const float ls = log_std[d];
    sigma[d] = __expf(ls);
    inv_s2[d] = 1.f / (sigma[d] * sigma[d]);
    base += ls;
  }

  const float inv_b = 1.f / (float)B;
  float loss_acc = 0.f;
  float dls[DT];
  #pragma unroll
  for (int d = 0; d < DT; ++d) dls[d] = 0.f;

  for (int r = gid; r < B; r += gridDim.x * LOSS_P_THREADS) {
    float diff[DT];
    float q = 0.f;
    #pragma unroll
    for (int d = 0; d < DT; ++d) {
      diff[d] = actions[(long)r * DT + d] - mean[(long)r * DT + d];
      const float z = diff[d] / sigma[d];
      q += z * z;
    }
    const float logp = -0.5f * q - base;
    float loss_r;
    const float c = dlogp_coeff(mode, logp, old_logp ? old_logp[r] : 0.f,
                                adv[r], clip, &loss_r) * inv_b;
    loss_acc += loss_r * inv_b;
    #pragma unroll
    for (int d = 0; d < DT; ++d) {
      dmean[(long)r * DT + d] = c * diff[d] * inv_s2[d];
      const float z2 = diff[d] * diff[d] * inv_s2[d];
      dls[d] += c * (z2 - 1.f);
    }
  }

  // per-block fixed-order reductions -> partials row
  #pragma unroll
  for (int d = 0; d < DT; ++d) {
    float v = wave_reduce_sum(dls[d]);
    if ((tid & 63) == 0) red[tid / WAVE] = v;
    __syncthreads();
    if (tid == 0) {
      float t = 0.f;
      for (int w = 0; w < LOSS_P_THREADS / WAVE; ++w) t += red[w];
      partials[(long)blockIdx.x * (DT + 1) + d] = t;
    }
    __syncthreads();
  }
  {
    float v = wave_reduce_sum(loss_acc);
    if ((tid & 63) == 0) red[tid / WAVE] = v;
    __syncthreads();
    if (tid == 0) {
      float t = 0.f;
      for (int w = 0; w < LOSS_P_THREADS / WAVE; ++w) t += red[w];
      partials[(long)blockIdx.x * (DT + 1) + DT] = t;
    }
  }
}

// Generic runtime-D variant (D in (8, GAUSS_MAX_D]): the per-thread
// dlog_std register accumulators of the template version don't exist at
// runtime D, so the block makes two passes over ITS OWN rows — phase A
// stashes each row's dlogp coefficient in c_buf and writes dmean, phase
// B loops d outermost and re-accumulates dlog_std per thread over the
// same rows.  Per-(thread,d) accumulation order over rows is IDENTICAL
// to the template version, so both are bitwise-equal and deterministic.
__global__ __launch_bounds__(LOSS_P_THREADS) void gaussian_policy_loss_bwd_g(
    const float* __restrict__ mean, const float* __restrict__ actions,
    const float* __restrict__ old_logp, const float* __restrict__ adv,
    const float* __restrict__ log_std, float* __restrict__ dmean,
    float* __restrict__ c_buf, float* __restrict__ partials, int B, int D,
    float clip, int mode) {
  __shared__ float red[LOSS_P_THREADS / WAVE];
  __shared__ float s_inv_s2[GAUSS_MAX_D];
  __shared__ float s_base;
  const int tid = threadIdx.x;
  for (int d = tid; d < D; d += LOSS_P_THREADS) {
    const float s = __expf(log_std[d]);
    s_inv_s2[d] = 1.f / (s * s);
  }
  if (tid == 0) {
    float b = 0.5f * (float)D * LOG_2PI;
    for (int d = 0; d < D; ++d) b += log_std[d];
    s_base = b;
  }
  __syncthreads();
  const float inv_b = 1.f / (float)B;
  float loss_acc = 0.f;
  for (int r = blockIdx.x * LOSS_P_THREADS + tid; r < B;
       r += gridDim.x * LOSS_P_THREADS) {
    float q = 0.f;
    for (int d = 0; d < D; ++d) {
      const float diff = actions[(long)r * D + d] - mean[(long)r * D + d];
      q += diff * diff * s_inv_s2[d];
    }
    const float logp = -0.5f * q - s_base;
    float loss_r;
    const float c = dlogp_coeff(mode, logp, old_logp ? old_logp[r] : 0.f,
                                adv[r], clip, &loss_r) * inv_b;
    c_buf[r] = c;
    loss_acc += loss_r * inv_b;
    for (int d = 0; d < D; ++d) {
      const float diff = actions[(long)r * D + d] - mean[(long)r * D + d];
      dmean[(long)r * D + d] = c * diff * s_inv_s2[d];
    }
  }
  __syncthreads();
  for (int d = 0; d < D; ++d) {
    float accd = 0.f;
    for (int r = blockIdx.x * LOSS_P_THREADS + tid; r < B;
         r += gridDim.x * LOSS_P_THREADS) {
      const float diff = actions[(long)r * D + d] - mean[(long)r * D + d];
      accd += c_buf[r] * (diff * diff * s_inv_s2[d] - 1.f);
    }
    float v = wave_reduce_sum(accd);
    if ((tid & 63) == 0) red[tid / WAVE] = v;
    __syncthreads();
    if (tid == 0) {
      float t = 0.f;
      for (int w = 0; w < LOSS_P_THREADS / WAVE; ++w) t += red[w];
      partials[(long)blockIdx.x * (D + 1) + d] = t;
    }
    __syncthreads();
  }
  {
    float v = wave_reduce_sum(loss_acc);
    if ((tid & 63) == 0) red[tid / WAVE] = v;
    __syncthreads();
    if (tid == 0) {
      float t = 0.f;
      for (int w = 0; w < LOSS_P_THREADS / WAVE; ++w) t += red[w];
      partials[(long)blockIdx.x * (D + 1) + D] = t;
    }
  }
}

// sum the per-block partial rows in fixed order:
// dlog_std[d] = sum_blk partials[blk][d]; scalars[0] = sum_blk partials[blk][D]
__global__ void loss_partials_finalize(const float* __restrict__ partials,
                                       float* __restrict__ dlog_std,
                                       float* __restrict__ scalars, int n_blocks,
                                       int D) {
  const int d = threadIdx.x;
  if (d > D) return;
  float s = 0.f;
  for (int p = 0; p < n_blocks; ++p) s += partials[(long)p * (D + 1) + d];
  if (d < D) dlog_std[d] = s;
  else scalars[0] = s;
}

// batched row finalize for the captured value loop: one thread per row,
// SAME serial summation order as loss_partials_finalize -> a deferred
// finalize is bitwise-identical to 80 per-iteration launches.
__global__ void value_loss_finalize_rows(const float* __restrict__ partials,
                                         float* __restrict__ scalars, int rows,
                                         int fb, int row_stride) {
  const int r = blockIdx.x * 256 + threadIdx.x;
  if (r >= rows) return;
  const float* row = partials + (long)r * row_stride;
  float s = 0.f;
  for (int p = 0; p < fb; ++p) s += row[p];
  scalars[r] = s;
}

// logp only (old-policy snapshot at epoch start)
__global__ __launch_bounds__(LOSS_THREADS) void gaussian_logp_kernel(
    const float* __restrict__ mean, const float* __restrict__ actions,
    const float* __restrict__ log_std, float* __restrict__ logp, int B, int D) {
  const int tid = threadIdx.x + blockIdx.x * LOSS_THREADS;
  __shared__ float s_sigma[GAUSS_MAX_D];
  __shared__ float s_base;
  for (int d = threadIdx.x; d < D; d += LOSS_THREADS)
    s_sigma[d] = __expf(log_std[d]);
  if (threadIdx.x == 0) {
    float b = 0.f;
    for (int d = 0; d < D; ++d) b += log_std[d];
    s_base = b + 0.5f * (float)D * LOG_2PI;
  }
  __syncthreads();
  for (int r = tid; r < B; r += gridDim.x * LOSS_THREADS) {
    float q = 0.f;
    for (int d = 0; d < D; ++d) {
      const float z = (actions[(long)r * D + d] - mean[(long)r * D + d]) / s_sigma[d];
      q += z * z;
    }
    logp[r] = -0.5f * q - s_base;
  }
}

// approx KL = mean(old_logp - logp(mean_new))  (ppo.py:259-269)
__global__ __launch_bounds__(LOSS_THREADS) void gaussian_kl_kernel(
    const float* __restrict__ mean, const float* __restrict__ actions,
    const float* __restrict__ log_std, const float* __restrict__ old_logp,
    float* __restrict__ out, int B, int D) {
  __shared__ float red[LOSS_THREADS / WAVE];
  __shared__ float s_sigma[GAUSS_MAX_D];
  const int tid = threadIdx.x;
  for (int d = tid; d < D; d += LOSS_THREADS) s_sigma[d] = __expf(log_std[d]);
  __syncthreads();
  float base = 0.f;
  for (int d = 0; d < D; ++d) base += log_std[d];
  base += 0.5f * (float)D * LOG_2PI;
  float acc = 0.f;
  for (int r = tid; r < B; r += LOSS_THREADS) {
    float q = 0.f;
    for (int d = 0; d < D; ++d) {
      const float z = (actions[(long)r * D + d] - mean[(long)r * D + d]) / s_sigma[d];
      q += z * z;
    }
    acc += old_logp[r] - (-0.5f * q - base);
  }
  const float total = block_sum(acc, red);
  if (tid == 0) out[0] = total / (float)B;
}

// ---------------------------------------------------------------------------
// Categorical policy (categorical_policy.py:22-32): logp = logit_a - lse
// outputs: dlogits[B,N], scalars[0]=loss
// ---------------------------------------------------------------------------
DEV_INLINE float row_lse(const float* logits, int n) {
  float m = logits[0];
  for (int j = 1; j < n; ++j) m = fmaxf(m, logits[j]);
  float s = 0.f;
  for (int j = 0; j < n; ++j) s += __expf(logits[j] - m);
  return m + __logf(s);
}

__global__ __launch_bounds__(LOSS_P_THREADS) void categorical_policy_loss_bwd(
    const float* __restrict__ logits, const float* __restrict__ actions,
    const float* __restrict__ old_logp, const float* __restrict__ adv,
    float* __restrict__ dlogits, float* __restrict__ partials, int B, int N,
    float clip, int mode) {
  __shared__ float red[LOSS_P_THREADS / WAVE];
  const int tid = threadIdx.x;
  const float inv_b = 1.f / (float)B;
  float loss_acc = 0.f;
  for (int r = blockIdx.x * LOSS_P_THREADS + tid; r < B;
       r += gridDim.x * LOSS_P_THREADS) {
    const float* lg = logits + (long)r * N;
    const int a = (int)actions[r];
    const float lse = row_lse(lg, N);
    const float logp = lg[a] - lse;
    float loss_r;
    const float c = dlogp_coeff(mode, logp, old_logp ? old_logp[r] : 0.f,
                                adv[r], clip, &loss_r) * inv_b;
    loss_acc += loss_r * inv_b;
    for (int j = 0; j < N; ++j) {
      const float p = __expf(lg[j] - lse);
      dlogits[(long)r * N + j] = c * (((j == a) ? 1.f : 0.f) - p);
    }
  }
  float v = wave_reduce_sum(loss_acc);
  if ((tid & 63) == 0) red[tid / WAVE] = v;
  __syncthreads();
  if (tid == 0) {
    float t = 0.f;
    for (int w = 0; w < LOSS_P_THREADS / WAVE; ++w) t += red[w];
    partials[blockIdx.x] = t;  // rows of size 1 (D=0) for the finalizer
  }
}

__global__ __launch_bounds__(LOSS_THREADS) void categorical_logp_kernel(
    const float* __restrict__ logits, const float* __restrict__ actions,
    float* __restrict__ logp, int B, int N) {
  const int tid = threadIdx.x + blockIdx.x * LOSS_THREADS;
  for (int r = tid; r < B; r += gridDim.x * LOSS_THREADS) {
    const float* lg = logits + (long)r * N;
    logp[r] = lg[(int)actions[r]] - row_lse(lg, N);
  }
}

__global__ __launch_bounds__(LOSS_THREADS) void categorical_kl_kernel(
    const float* __restrict__ logits, const float* __restrict__ actions,
    const float* __restrict__ old_logp, float* __restrict__ out, int B, int N) {
  __shared__ float red[LOSS_THREADS / WAVE];
  const int tid = threadIdx.x;
  float acc = 0.f;
  for (int r = tid; r < B; r += LOSS_THREADS) {
    const float* lg = logits + (long)r * N;
    acc += old_logp[r] - (lg[(int)actions[r]] - row_lse(lg, N));
  }
  const float total = block_sum(acc, red);
  if (tid == 0) out[0] = total / (float)B;
}

// host-side dispatch: compile-time-D instantiations for the common
// D <= 8 heads (register dlog_std accumulators), generic two-pass
// kernel for D in (8, GAUSS_MAX_D] (c_buf: caller-provided [B] scratch)
void launch_gaussian_loss(const float* mean, const float* actions,
                          const float* old_logp, const float* adv,
                          const float* log_std, float* dmean, float* c_buf,
                          float* partials, int B, int D, float clip, int mode,
                          int n_blocks, hipStream_t stream) {
  dim3 g(n_blocks), b(LOSS_P_THREADS);
#define CASE_D(DT)                                                           \
  case DT:                                                                   \
    hipLaunchKernelGGL((gaussian_policy_loss_bwd_t<DT>), g, b, 0, stream,    \
                       mean, actions, old_logp, adv, log_std, dmean,         \
                       partials, B, clip, mode);                             \
    break;
  switch (D) {
    CASE_D(1) CASE_D(2) CASE_D(3) CASE_D(4)
    CASE_D(5) CASE_D(6) CASE_D(7) CASE_D(8)
    default:
      hipLaunchKernelGGL(gaussian_policy_loss_bwd_g, g, b, 0, stream, mean,
                         actions, old_logp, adv, log_std, dmean, c_buf,
                         partials, B, D, clip, mode);
      break;
  }
#undef CASE_D
}

// ---------------------------------------------------------------------------
// value MSE: loss = mean((v - ret)^2), dv = 2(v - ret)/B  (ppo.py:283-287)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(LOSS_THREADS) void value_mse_bwd_kernel(
    const float* __restrict__ v, const float* __restrict__ ret,
    float* __restrict__ dv, float* __restrict__ scalars, int B) {
  __shared__ float red[LOSS_THREADS / WAVE];
  const int tid = threadIdx.x;
  const float inv_b = 1.f / (float)B;
  float loss_acc = 0.f;
  for (int r = tid; r < B; r += LOSS_THREADS) {
    const float diff = v[r] - ret[r];
    loss_acc += diff * diff * inv_b;
    dv[r] = 2.f * diff * inv_b;
  }
  const float loss = block_sum(loss_acc, red);
  if (tid == 0) scalars[0] = loss;
}
