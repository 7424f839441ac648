// Multi-tensor parameter-update kernels: fused Adam and fused Polyak.
//
// The reference updates parameters through torch.optim.Adam and a
// per-parameter Python Polyak loop (utils.py:47-57); at this model
// scale (3-100 K params over ~7 tensors) those cost one-plus kernel
// launch per tensor.  Here the whole update is ONE launch: the tensor
// table travels by value in the kernarg segment (no pointer-table
// upload), one workgroup per tensor, grid-stride within.
#include "common.h"

__global__ __launch_bounds__(256) void fused_adam_kernel(AdamArgs a) {
  const int t = blockIdx.x;
  if (t >= a.n_tensors) return;
  const int n = a.numel[t];
  const float s = a.step[t][0] + 1.f;
  const float bc1 = 1.f - __powf(a.beta1, s);
  const float bc2 = 1.f - __powf(a.beta2, s);

  float* p = a.p[t];
  float* g = a.g[t];
  float* m = a.m[t];
  float* v = a.v[t];
  for (int i = threadIdx.x; i < n; i += 256) {
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
  __syncthreads();
  if (threadIdx.x == 0) a.step[t][0] = s;
}

__global__ __launch_bounds__(256) void fused_polyak_kernel(PolyakArgs a) {
  const int t = blockIdx.x;
  if (t >= a.n_tensors) return;
  const int n = a.numel[t];
  const float* s = a.src[t];
  float* d = a.dst[t];
  const float rho = a.rho, one_m = 1.f - a.rho;
  for (int i = threadIdx.x; i < n; i += 256) {
    d[i] = rho * d[i] + one_m * s[i];
  }
}
