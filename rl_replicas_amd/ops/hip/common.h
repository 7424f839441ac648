// Shared device helpers for the rl_replicas_amd CDNA4 (gfx950) kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef __hip_bfloat16 bf16_t;
// bf16 MFMA fragment: 8 elements per lane (v_mfma_f32_16x16x32_bf16)
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

DEV_INLINE __bf16 f32_to_bf16(float v) {
  return (__bf16)v;  // hardware RNE conversion on gfx950
}

// activation codes shared with python (ops/fused_mlp.py)
#define ACT_IDENTITY 0
#define ACT_TANH 1
#define ACT_RELU 2

DEV_INLINE float act_apply(int code, float x) {
  switch (code) {
    case ACT_TANH: return tanhf(x);
    case ACT_RELU: return x > 0.f ? x : 0.f;
    default: return x;
  }
}

// derivative of the activation expressed through the POST-activation y
// (tanh' = 1 - y^2, relu' = y > 0), so backward only needs saved outputs
DEV_INLINE float act_grad_from_y(int code, float y) {
  switch (code) {
    case ACT_TANH: return 1.f - y * y;
    case ACT_RELU: return y > 0.f ? 1.f : 0.f;
    default: return 1.f;
  }
}

// PPO/VPG gradient coefficient wrt logp for one row (caller scales by
// 1/B):  mode 0 = VPG (-A), mode 1 = PPO clipped surrogate.  Gradient
// semantics replicate torch exactly (min-tie 0.5/0.5, clamp-inclusive).
#define LOG_2PI 1.8378770664093453f
DEV_INLINE float dlogp_coeff(int mode, float logp, float old_logp, float adv,
                             float clip, float* loss_out) {
  if (mode == 0) {
    *loss_out = -logp * adv;
    return -adv;
  }
  const float ratio = __expf(logp - old_logp);
  const float lo = 1.f - clip, hi = 1.f + clip;
  const float rc = fminf(fmaxf(ratio, lo), hi);
  const float s1 = ratio * adv;
  const float s2 = rc * adv;
  *loss_out = -fminf(s1, s2);
  const float g1 = (s1 < s2) ? 1.f : (s1 == s2 ? 0.5f : 0.f);
  const float inclip = (ratio >= lo && ratio <= hi) ? 1.f : 0.f;
  return -(g1 * s1 + (1.f - g1) * inclip * s1);
}

// wave-level f32 sum (64 lanes)
DEV_INLINE float wave_reduce_sum(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

// Fused-MLP static limits (mirrored in ops/fused_mlp.py)
#define MLP_MAX_LAYERS 5
#define MLP_MAX_WIDTH 256
#define MLP_ROWS 64            // rows per workgroup (4 waves x 16)
#define MLP_LDSW (MLP_MAX_WIDTH + 4)  // padded LDS row stride (bank spread)

// kernel-argument block for the fused MLP kernels (passed by value)
struct MLPArgs {
  const float* w[MLP_MAX_LAYERS];
  const float* b[MLP_MAX_LAYERS];
  float* h[MLP_MAX_LAYERS];  // saved post-activation outputs (h[L-1] = final out)
  int dims[MLP_MAX_LAYERS + 1];
  int acts[MLP_MAX_LAYERS];
  int n_layers;
  int batch;
};

// whole-net fused-backward argument block (mlp_kernels.hip)
struct MLPBwdArgs {
  const float* w[MLP_MAX_LAYERS];
  const float* h[MLP_MAX_LAYERS];  // post-activations; h[L-1] = final out
  const float* b[MLP_MAX_LAYERS];  // biases (DO_FWD mode only)
  int dims[MLP_MAX_LAYERS + 1];
  int acts[MLP_MAX_LAYERS];
  int n_layers;
  int batch;
  long ws_stride;
  int layer_off[MLP_MAX_LAYERS];  // flat elem offset of each layer's partials
};

// all-layer gradient-reduction argument block (mlp_kernels.hip).
// workspace layout: ws[partial][flat-elem] with the layers' (od*id+od)
// segments concatenated in layer order inside each partial row.
struct ReduceAllArgs {
  const float* ws;            // [n_blocks][stride]
  long stride;                // flat elems per partial row (grand total)
  float* dw[MLP_MAX_LAYERS];
  float* db[MLP_MAX_LAYERS];
  int total[MLP_MAX_LAYERS];  // od*id + od per layer
  int wsize[MLP_MAX_LAYERS];  // od*id (split point)
  int n_layers;
  int n_blocks;
};

// multi-tensor update argument blocks (update_kernels.hip / bindings.hip)
#define MT_MAX_TENSORS 16

struct AdamArgs {
  float* p[MT_MAX_TENSORS];
  float* g[MT_MAX_TENSORS];
  float* m[MT_MAX_TENSORS];
  float* v[MT_MAX_TENSORS];
  float* step[MT_MAX_TENSORS];  // scalar step counters (fp32, torch-compatible)
  int numel[MT_MAX_TENSORS];
  int n_tensors;
  float lr, beta1, beta2, eps, weight_decay;
  // added to step[t] when computing bias correction (captured value loop
  // bakes per-iteration deltas so ONE bump per graph suffices); also the
  // bump amount for adam_step_bump_kernel
  float step_delta;
  // optional device gate: update and bump are skipped while *gate == 0
  // (captured PPO policy loop: KL early stop without host control flow)
  const float* gate;
};

struct PolyakArgs {
  const float* src[MT_MAX_TENSORS];
  float* dst[MT_MAX_TENSORS];
  int numel[MT_MAX_TENSORS];
  int n_tensors;
  float rho;
};

// merged gradient-reduce + Adam update (mlp_kernels.hip): consumes the
// backward stage-1 workspace and applies the optimizer in the SAME
// kernel — no dw/db round trip through HBM, one fewer dependent launch
// per optimizer step in the captured loops
struct ReduceAdamArgs {
  const float* ws;  // [n_blocks][stride] partials
  long stride;
  float* pw[MLP_MAX_LAYERS];
  float* pb[MLP_MAX_LAYERS];
  float* mw[MLP_MAX_LAYERS];
  float* mb[MLP_MAX_LAYERS];
  float* vw[MLP_MAX_LAYERS];
  float* vb[MLP_MAX_LAYERS];
  const float* step;  // shared pre-bump step counter (params step in lockstep)
  int total[MLP_MAX_LAYERS];
  int wsize[MLP_MAX_LAYERS];
  int n_layers, n_blocks;
  float lr, beta1, beta2, eps, weight_decay, step_delta;
  const float* gate;  // optional: skip entirely while *gate == 0
};

// Gaussian-PPO seed for the DO_FWD fused backward (mlp_kernels.hip):
// when `actions` is non-null the dZ seed of the last (identity-head)
// layer is the PPO clipped-surrogate gradient computed from the
// LDS-resident forward output, and the kernel additionally emits
// per-block loss partials, pending-KL partials, and dlog_std partials
// (written into the stage-1 workspace at dls_off as a pseudo-layer).
struct GaussSeedArgs {
  const float* actions;   // [B, D]; nullptr -> plain dy/mse seed
  const float* old_logp;  // [B]
  const float* adv;       // [B]
  const float* log_std;   // [D]
  float* kl_partials;     // [n_blocks]
  int dls_off;            // flat ws offset of the dlog_std pseudo-layer
  float clip;
};

// replay-ring minibatch gather (offpolicy_kernels.hip / bindings.hip)
struct ReplayGatherArgs {
  const float* obs;   // [cap, O] ring storage
  const float* act;   // [cap, A]
  const float* rew;   // [cap]
  const float* nxt;   // [cap, O]
  const float* dn;    // [cap]
  const long long* size;  // device scalar: current ring fill
  float* qin;      // [B, O+A]  critic input (obs|act)
  float* obs_out;  // [B, O]    actor input
  float* nxt_out;  // [B, O]
  float* rew_out;  // [B]
  float* dn_out;   // [B]
  int B, O, A;
  uint64_t seed, offset;
  const unsigned long long* offset_ptr;  // optional device RNG counter
};
