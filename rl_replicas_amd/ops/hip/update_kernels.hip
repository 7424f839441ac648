// Multi-tensor parameter-update kernels: fused Adam and fused Polyak.
//
// The reference updates parameters through torch.optim.Adam and a
// per-parameter Python Polyak loop (utils.py:47-57); at this model
// scale those cost one-plus kernel launch per tensor.  Here the whole
// update is ONE launch (plus a one-block step-counter bump for Adam):
// the tensor table travels by value in the kernarg segment, and the
// grid is 2-D — tensors x element chunks — so the 65K-element tensors
// of the off-policy nets update across many CUs instead of one
// workgroup (one-block-per-tensor measured 114 us; chunked ~10 us).
//
// The step counters are read (s+1) by every block and bumped by a
// separate single-block kernel AFTER the update — concurrent blocks
// must all see the same pre-update count.
#include "common.h"

#define MT_CHUNK 8192  // elements per (tensor, chunk) block

__global__ __launch_bounds__(256) void fused_adam_kernel(AdamArgs a) {
  // device-side gate: a captured loop can disable later updates without
  // host control flow (PPO's KL early stop runs entirely on device)
  if (a.gate && *a.gate == 0.f) return;
  const int t = blockIdx.x;
  if (t >= a.n_tensors) return;
  const int n = a.numel[t];
  const int c0 = blockIdx.y * MT_CHUNK;
  if (c0 >= n) return;
  const int c1 = min(c0 + MT_CHUNK, n);
  const float s = a.step[t][0] + 1.f + a.step_delta;
  const float bc1 = 1.f - __powf(a.beta1, s);
  const float bc2 = 1.f - __powf(a.beta2, s);

  float* p = a.p[t];
  float* g = a.g[t];
  float* m = a.m[t];
  float* v = a.v[t];
  for (int i = c0 + threadIdx.x; i < c1; i += 256) {
    float grad = g[i];
    if (a.weight_decay != 0.f) grad += a.weight_decay * p[i];
    float mi = a.beta1 * m[i] + (1.f - a.beta1) * grad;
    float vi = a.beta2 * v[i] + (1.f - a.beta2) * grad * grad;
    m[i] = mi;
    v[i] = vi;
    const float mhat = mi / bc1;
    const float vhat = vi / bc2;
    p[i] -= a.lr * mhat / (sqrtf(vhat) + a.eps);
  }
}

// bump every tensor's step counter once per optimizer step (launched
// after fused_adam_kernel on the same stream)
__global__ void adam_step_bump_kernel(AdamArgs a) {
  if (a.gate && *a.gate == 0.f) return;
  const int t = threadIdx.x;
  if (t < a.n_tensors) a.step[t][0] += a.step_delta;  // 1.0 for a plain step
}

// bump by a DEVICE scalar (captured gated loops: the executed-iteration
// count is only known on device)
__global__ void adam_step_bump_dev_kernel(AdamArgs a,
                                          const float* __restrict__ amount) {
  const int t = threadIdx.x;
  if (t < a.n_tensors) a.step[t][0] += *amount;
}

// Gate bookkeeping fed by the mega policy-iteration kernel's per-block
// partials: sums loss and pending-KL partials (fixed order), optionally
// records the iteration-0 loss, then applies the gate rule.  first_iter
// has no pending KL (old == current params), so it only stores the loss.
__global__ void ppo_gate_update_reduce_kernel(
    float* gate, const float* kl_partials, const float* loss_partials,
    int n_blocks, float inv_b, float* kl_final, float* iters_done, float thr,
    float* loss_out, int first_iter) {
  if (threadIdx.x != 0) return;
  float sk = 0.f, sl = 0.f;
  for (int p = 0; p < n_blocks; ++p) {
    sk += kl_partials[p];
    sl += loss_partials[p];
  }
  if (first_iter) {
    *loss_out = sl;
    return;  // no pending KL at iteration 0
  }
  const float kl = sk * inv_b;
  if (*gate != 0.f) {
    *kl_final = kl;
    *iters_done += 1.f;
    if (kl > thr) *gate = 0.f;
  }
}

// One-thread bookkeeping for the captured PPO policy loop's device-side
// KL early stop: while the gate is open, record the KL and iteration
// count; close the gate when the KL crosses the threshold.  Replaces a
// ~6-torch-op chain per captured iteration.
__global__ void ppo_gate_update_kernel(float* gate, const float* kl,
                                       float* kl_final, float* iters_done,
                                       float thr) {
  if (*gate != 0.f) {
    *kl_final = *kl;
    *iters_done += 1.f;
    if (*kl > thr) *gate = 0.f;
  }
}

__global__ __launch_bounds__(256) void fused_polyak_kernel(PolyakArgs a) {
  const int t = blockIdx.x;
  if (t >= a.n_tensors) return;
  const int n = a.numel[t];
  const int c0 = blockIdx.y * MT_CHUNK;
  if (c0 >= n) return;
  const int c1 = min(c0 + MT_CHUNK, n);
  const float* s = a.src[t];
  float* d = a.dst[t];
  const float rho = a.rho, one_m = 1.f - a.rho;
  for (int i = c0 + threadIdx.x; i < c1; i += 256) {
    d[i] = rho * d[i] + one_m * s[i];
  }
}
