// PyTorch bindings for the rl_replicas_amd HIP/CDNA4 kernels.
//
// Built in-tree as rl_replicas_amd/_hip_ops*.so (setup.py,
// PYTORCH_ROCM_ARCH=gfx950).  All entry points run on the current
// torch HIP stream and validate device/dtype/contiguity up front so a
// misuse fails loudly instead of silently falling back.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <vector>

#include "common.h"

// kernel declarations (defined in the .hip translation units)
void launch_mlp_fwd(const MLPArgs& args, const float* x, int save_hidden,
                    int rows, int maxw, int n_blocks, int wstage_mode,
                    size_t lds_bytes, int compute_bf16, hipStream_t stream);
void launch_mlp_bwd_layer(const float* dy, const float* y, const float* xin,
                          const float* W, float* dx, float* ws, long ws_stride,
                          int batch, int out_d, int in_d, int act, int rows,
                          int maxw, int n_blocks, int wstage_mode,
                          size_t lds_bytes, hipStream_t stream);
void launch_mlp_bwd_fused(const MLPBwdArgs& args, const float* x,
                          const float* dy, float* dx, float* ws,
                          const float* mse_returns, float* loss_partials,
                          size_t lds_bytes, int n_blocks, int compute_bf16,
                          hipStream_t stream, int rows, int do_fwd = 0,
                          GaussSeedArgs gargs = GaussSeedArgs{});
// Backward tiles stay at 32 rows: 16-row tiles made stage-1 faster but
// DOUBLED the partial-row count, pushing the (latency-bound) reduce
// from 10.3 to 17.8 us — a measured net loss.  (The FORWARD still uses
// 16-row tiles below 8K rows; its cost has no reduce to pay.)
inline int bwd_fused_rows(int batch) { (void)batch; return 32; }
__global__ void mlp_grad_reduce_onepass_f32(ReduceAllArgs a);
__global__ void mlp_grad_reduce_adam_f32(ReduceAdamArgs a);
void launch_mlp_layer_fwd_wide(const float* x, const float* W, const float* B,
                               float* out, int batch, int in_d, int out_d,
                               int act, hipStream_t stream);
void launch_mlp_bwd_wide(const float* dy, const float* y, const float* xin,
                         const float* W, float* dx, float* ws, long ws_stride,
                         int batch, int out_d, int in_d, int act,
                         hipStream_t stream);
void launch_mlp_dgrad_wide(const float* dy, const float* y, const float* W,
                           float* dx, int batch, int out_d, int in_d, int act,
                           hipStream_t stream);
void launch_gaussian_loss(const float* mean, const float* actions,
                          const float* old_logp, const float* adv,
                          const float* log_std, float* dmean, float* c_buf,
                          float* partials, int B, int D, float clip, int mode,
                          int n_blocks, hipStream_t stream);
__global__ void loss_partials_finalize(const float* partials, float* dlog_std,
                                       float* scalars, int n_blocks, int D);
__global__ void gaussian_logp_kernel(const float* mean, const float* actions,
                                     const float* log_std, float* logp, int B,
                                     int D);
__global__ void gaussian_kl_kernel(const float* mean, const float* actions,
                                   const float* log_std, const float* old_logp,
                                   float* out, int B, int D);
__global__ void categorical_policy_loss_bwd(const float* logits,
                                            const float* actions,
                                            const float* old_logp,
                                            const float* adv, float* dlogits,
                                            float* partials, int B, int N,
                                            float clip, int mode);
__global__ void categorical_logp_kernel(const float* logits, const float* actions,
                                        float* logp, int B, int N);
__global__ void categorical_kl_kernel(const float* logits, const float* actions,
                                      const float* old_logp, float* out, int B,
                                      int N);
__global__ void value_loss_finalize_rows(const float* partials, float* scalars,
                                         int rows, int fb, int row_stride);
__global__ void value_mse_bwd_kernel(const float* v, const float* ret, float* dv,
                                     float* scalars, int B);
__global__ void gaussian_sample_kernel(const float* mean, const float* log_std,
                                       float* out, int B, int D, uint64_t seed,
                                       uint64_t offset, float noise_scale,
                                       float limit,
                                       const unsigned long long* offset_ptr);
__global__ void categorical_sample_kernel(const float* logits, int64_t* out,
                                          int B, int N, uint64_t seed,
                                          uint64_t offset);
__global__ void synthetic_env_step_kernel(const float* state, const float* actions,
                                          const float* A, const float* Bm,
                                          const float* w, float* s_out,
                                          float* final_out, float* reward, int N,
                                          int O, int Adim, float sigma,
                                          uint64_t seed, uint64_t offset,
                                          int do_reset,
                                          const unsigned long long* offset_ptr);
__global__ void synthetic_env_reset_kernel(float* s_out, int total,
                                           uint64_t seed, uint64_t offset,
                                           const unsigned long long* offset_ptr);
__global__ void counter_add_kernel(unsigned long long* ctr,
                                   unsigned long long delta);
__global__ void segmented_gae_kernel(const float* rewards, const float* values,
                                     const float* last_values, const int* offsets,
                                     const int* dones, float* advantages,
                                     float* returns, float gamma, float lam);
__global__ void normalize_kernel(const float* x, float* y, int n);
__global__ void q_target_kernel(const float* r, const float* d, const float* qn,
                                float* out, float gamma, int n);
__global__ void td3_smooth_kernel(const float* a, float* out, int total,
                                  uint64_t seed, uint64_t offset, float scale,
                                  float clip, float limit,
                                  const unsigned long long* offset_ptr);
__global__ void q_target_min2_kernel(const float* r, const float* d,
                                     const float* q1, const float* q2,
                                     float* out, float gamma, int n);
__global__ void replay_gather_kernel(ReplayGatherArgs a);

__global__ void fused_adam_kernel(AdamArgs a);
__global__ void adam_step_bump_kernel(AdamArgs a);
__global__ void fused_polyak_kernel(PolyakArgs a);
__global__ void ppo_gate_update_kernel(float* gate, const float* kl,
                                       float* kl_final, float* iters_done,
                                       float thr);
__global__ void ppo_gate_update_reduce_kernel(
    float* gate, const float* kl_partials, const float* loss_partials,
    int n_blocks, float inv_b, float* kl_final, float* iters_done, float thr,
    float* loss_out, int first_iter);

namespace {

hipStream_t current_stream() { return c10::hip::getCurrentHIPStream().stream(); }

void check_f32_gpu(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

#define HIP_OK(expr)                                                        \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));    \
  } while (0)

// merged reduce+Adam launch (optional fast path of mlp_backward /
// value_mlp_backward): m/v are ordered [weights..., biases...]
void launch_reduce_adam(const std::vector<torch::Tensor>& weights,
                        const std::vector<torch::Tensor>& biases,
                        const std::vector<int64_t>& totals, const float* ws,
                        long stride, int n_blocks, int64_t grand,
                        const std::vector<torch::Tensor>& m,
                        const std::vector<torch::Tensor>& v,
                        const torch::Tensor& step, double lr, double beta1,
                        double beta2, double eps, double weight_decay,
                        double step_delta,
                        const c10::optional<torch::Tensor>& gate,
                        hipStream_t stream) {
  const int L = (int)weights.size();
  TORCH_CHECK((int)m.size() == 2 * L && (int)v.size() == 2 * L,
              "adam m/v must be [weights..., biases...] (2L tensors)");
  ReduceAdamArgs ra{};
  ra.ws = ws;
  ra.stride = stride;
  ra.n_layers = L;
  ra.n_blocks = n_blocks;
  for (int l = 0; l < L; ++l) {
    ra.pw[l] = weights[l].data_ptr<float>();
    ra.pb[l] = biases[l].data_ptr<float>();
    ra.mw[l] = m[l].data_ptr<float>();
    ra.mb[l] = m[L + l].data_ptr<float>();
    ra.vw[l] = v[l].data_ptr<float>();
    ra.vb[l] = v[L + l].data_ptr<float>();
    ra.total[l] = (int)totals[l];
    ra.wsize[l] = (int)(weights[l].size(0) * weights[l].size(1));
  }
  ra.step = step.data_ptr<float>();
  ra.lr = (float)lr;
  ra.beta1 = (float)beta1;
  ra.beta2 = (float)beta2;
  ra.eps = (float)eps;
  ra.weight_decay = (float)weight_decay;
  ra.step_delta = (float)step_delta;
  ra.gate = gate.has_value() ? gate->data_ptr<float>() : nullptr;
  int rb = (int)std::min<int64_t>(256, (grand + 63) / 64);
  hipLaunchKernelGGL(mlp_grad_reduce_adam_f32, dim3(rb), dim3(256), 0, stream,
                     ra);
  HIP_OK(hipGetLastError());
}

// (ROWS, MAXW) tile selection for the templated MLP kernels.
// MAXW: smallest instantiated width bound that fits every layer ->
// smaller LDS footprint, higher per-CU occupancy for the narrow
// on-policy nets.  ROWS: 32 at large batches fills the 256-CU chip
// (4000-row batch -> 125 workgroups); 64 otherwise.
void pick_tile(int batch, int max_width, int* rows, int* maxw) {
  *maxw = max_width <= 64 ? 64 : 256;
  // narrow nets: 16-row tiles up to 8K rows (4000-row rollout -> 250
  // WGs; measured +2.4% over 32-row on the PPO bench — latency hiding
  // beats weight-staging amortization until the chip is well past
  // full); 32-row beyond
  *rows = (*maxw == 256) ? 32 : (batch < 8192 ? 16 : 32);
  static int env_rows = []() {
    const char* e = getenv("RL_REPLICAS_AMD_MLP_ROWS");
    return e ? atoi(e) : 0;
  }();
  if (env_rows == 16 || env_rows == 32 || env_rows == 64) *rows = env_rows;
  if (*maxw == 256) *rows = 32;  // only <32,256> is instantiated
}

#define KCHUNK 64

// weight-staging mode + dynamic-LDS size for the fused MLP kernels:
// mode 0 stages the whole net per block (narrow nets), mode 1 stages
// per-wave W sub-tiles.  act_elems = the two activation ping-pong
// buffers (fwd) or dz+xt (bwd) — same footprint.
void pick_wstage(const std::vector<torch::Tensor>& weights, int rows, int maxw,
                 bool backward_single_layer, int layer, int* mode,
                 size_t* lds_bytes) {
  const int ldsw = maxw + 4;
  size_t act_elems = (size_t)2 * rows * ldsw;
  size_t whole = 0;
  if (backward_single_layer) {
    whole = (size_t)weights[layer].size(0) * (weights[layer].size(1) + 1);
  } else {
    for (auto& w : weights) whole += (size_t)w.size(0) * (w.size(1) + 1);
  }
  size_t mode0_bytes = (act_elems + whole) * 4;
  if (mode0_bytes <= 100 * 1024) {
    *mode = 0;
    *lds_bytes = mode0_bytes;
  } else {
    // wide nets: direct global W reads (mode 2) measure faster than the
    // serialized per-wave chunk staging (mode 1) at these grid sizes
    *mode = 2;
    *lds_bytes = act_elems * 4;
  }
  static int env_mode = []() {
    const char* e = getenv("RL_REPLICAS_AMD_WSTAGE");
    return e ? atoi(e) : -1;
  }();
  if (env_mode == 1 && *mode == 2) {
    *mode = 1;
    size_t wv = backward_single_layer ? (size_t)4 * KCHUNK * 18
                                      : (size_t)4 * 16 * (KCHUNK + 2);
    *lds_bytes = (act_elems + wv) * 4;
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// fused MLP
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> mlp_forward(torch::Tensor x,
                                       std::vector<torch::Tensor> weights,
                                       std::vector<torch::Tensor> biases,
                                       std::vector<int64_t> acts,
                                       bool save_hidden,
                                       int64_t compute_bf16) {
  const int L = (int)weights.size();
  TORCH_CHECK(L >= 1 && L <= MLP_MAX_LAYERS, "unsupported layer count ", L);
  check_f32_gpu(x, "x");
  TORCH_CHECK(x.dim() == 2, "x must be 2-D");

  MLPArgs args{};
  args.n_layers = L;
  args.batch = (int)x.size(0);
  args.dims[0] = (int)x.size(1);
  for (int l = 0; l < L; ++l) {
    check_f32_gpu(weights[l], "weight");
    check_f32_gpu(biases[l], "bias");
    args.dims[l + 1] = (int)weights[l].size(0);
    TORCH_CHECK((int)weights[l].size(1) == args.dims[l], "weight shape mismatch");
    TORCH_CHECK(args.dims[l + 1] <= MLP_MAX_WIDTH && args.dims[l] <= MLP_MAX_WIDTH,
                "layer width exceeds MLP_MAX_WIDTH");
    args.w[l] = weights[l].data_ptr<float>();
    args.b[l] = biases[l].data_ptr<float>();
    args.acts[l] = (int)acts[l];
  }

  int max_width = args.dims[0];
  for (int l = 1; l <= L; ++l) max_width = std::max(max_width, args.dims[l]);
  int rows, maxw;
  pick_tile(args.batch, max_width, &rows, &maxw);

  auto opts = x.options();
  // per-layer column-split kernels for wide nets at ANY batch: the
  // whole-net <32,256> path was measured SLOWER at minibatch scale
  // (batch 100 -> 4 WGs serializing 3 layers each, 800 vs 1232
  // env-steps/s on the TD3 bench) — concurrency across independent
  // per-layer launches beats fewer launches here
  const bool wide = maxw == 256;
  std::vector<torch::Tensor> outs;  // [final, h0..h_{L-2}]
  torch::Tensor final_out = torch::empty({x.size(0), args.dims[L]}, opts);
  outs.push_back(final_out);
  std::vector<torch::Tensor> inter;  // wide path needs intermediates always
  for (int l = 0; l < L - 1; ++l) {
    if (save_hidden || wide) {
      torch::Tensor h = torch::empty({x.size(0), args.dims[l + 1]}, opts);
      args.h[l] = h.data_ptr<float>();
      if (save_hidden) outs.push_back(h);
      else inter.push_back(h);
    } else {
      args.h[l] = nullptr;
    }
  }
  args.h[L - 1] = final_out.data_ptr<float>();

  if (args.batch > 0) {
    auto stream = current_stream();
    if (wide) {
      // per-layer 2D-grid kernels: column groups spread across blocks
      const float* cur = x.data_ptr<float>();
      for (int l = 0; l < L; ++l) {
        launch_mlp_layer_fwd_wide(cur, args.w[l], args.b[l], args.h[l],
                                  args.batch, args.dims[l], args.dims[l + 1],
                                  args.acts[l], stream);
        HIP_OK(hipGetLastError());
        cur = args.h[l];
      }
    } else {
      const int n_blocks = (args.batch + rows - 1) / rows;
      int wmode;
      size_t lds_bytes;
      pick_wstage(weights, rows, maxw, false, 0, &wmode, &lds_bytes);
      launch_mlp_fwd(args, x.data_ptr<float>(), save_hidden ? 1 : 0, rows, maxw,
                     n_blocks, wmode, lds_bytes, (int)compute_bf16, stream);
      HIP_OK(hipGetLastError());
    }
  }
  return outs;
}

std::vector<torch::Tensor> mlp_backward(torch::Tensor grad_out, torch::Tensor x,
                                        std::vector<torch::Tensor> weights,
                                        std::vector<torch::Tensor> biases,
                                        std::vector<torch::Tensor> hidden,
                                        torch::Tensor final_out,
                                        std::vector<int64_t> acts,
                                        int64_t compute_bf16,
                                        c10::optional<std::vector<torch::Tensor>> adam_m,
                                        c10::optional<std::vector<torch::Tensor>> adam_v,
                                        c10::optional<torch::Tensor> adam_step,
                                        double lr, double beta1, double beta2,
                                        double eps, double weight_decay,
                                        double adam_step_delta,
                                        c10::optional<torch::Tensor> adam_gate,
                                        bool input_grad_only) {
  const bool fuse_adam = adam_m.has_value();
  const int L = (int)weights.size();
  check_f32_gpu(grad_out, "grad_out");
  check_f32_gpu(x, "x");
  const int batch = (int)x.size(0);
  int max_width = (int)x.size(1);
  for (int l = 0; l < L; ++l)
    max_width = std::max(max_width, (int)weights[l].size(0));
  int rows, maxw;
  pick_tile(batch, max_width, &rows, &maxw);
  if (rows < 32) rows = 32;  // backward kernels are instantiated at 32/64
  const int n_blocks = (batch + rows - 1) / rows;
  auto opts = x.options();
  auto stream = current_stream();

  // one workspace: [block][concatenated layer elems] (layer-blind stage-1)
  std::vector<int64_t> totals(L), layer_off(L);
  int64_t grand = 0;
  for (int l = 0; l < L; ++l) {
    const int out_d = (int)weights[l].size(0);
    const int in_d = (int)weights[l].size(1);
    totals[l] = (int64_t)out_d * in_d + out_d;
    layer_off[l] = grand;
    grand += totals[l];
  }
  // whole-net fused backward for narrow nets when the LDS image fits
  size_t whole_w = 0;
  for (auto& w : weights) whole_w += (size_t)w.size(0) * (w.size(1) + 1);
  const int brows = bwd_fused_rows(batch);
  const size_t fused_lds = ((size_t)3 * brows * 68 + whole_w) * 4;
  const bool use_fused_bwd = (maxw == 64) && fused_lds <= 100 * 1024;
  if (use_fused_bwd) {
    const int fb = (batch + brows - 1) / brows;
    torch::Tensor ws = torch::empty({(int64_t)fb, grand}, opts);
    std::vector<torch::Tensor> dws(L), dbs(L);
    MLPBwdArgs ba{};
    ba.n_layers = L;
    ba.batch = batch;
    ba.ws_stride = grand;
    ba.dims[0] = (int)x.size(1);
    for (int l = 0; l < L; ++l) {
      ba.w[l] = weights[l].data_ptr<float>();
      ba.h[l] = (l == L - 1 ? final_out : hidden[l]).data_ptr<float>();
      ba.dims[l + 1] = (int)weights[l].size(0);
      ba.acts[l] = (int)acts[l];
      ba.layer_off[l] = (int)layer_off[l];
      dws[l] = torch::empty({(int)weights[l].size(0), (int)weights[l].size(1)}, opts);
      dbs[l] = torch::empty({(int)weights[l].size(0)}, opts);
    }
    torch::Tensor dx = torch::empty({batch, x.size(1)}, opts);
    launch_mlp_bwd_fused(ba, x.data_ptr<float>(),
                         grad_out.contiguous().data_ptr<float>(),
                         dx.data_ptr<float>(), ws.data_ptr<float>(), nullptr,
                         nullptr, fused_lds, fb, (int)compute_bf16, stream,
                         brows);
    HIP_OK(hipGetLastError());

    if (input_grad_only) return {dx};
    if (fuse_adam) {
      launch_reduce_adam(weights, biases, totals, ws.data_ptr<float>(), grand,
                         fb, grand, *adam_m, *adam_v, *adam_step, lr, beta1,
                         beta2, eps, weight_decay, adam_step_delta, adam_gate,
                         stream);
      return {dx};
    }
    ReduceAllArgs ra{};
    ra.ws = ws.data_ptr<float>();
    ra.stride = grand;
    ra.n_layers = L;
    ra.n_blocks = fb;
    for (int l = 0; l < L; ++l) {
      ra.dw[l] = dws[l].data_ptr<float>();
      ra.db[l] = dbs[l].data_ptr<float>();
      ra.total[l] = (int)totals[l];
      ra.wsize[l] = (int)(weights[l].size(0) * weights[l].size(1));
    }
    int rb = (int)std::min<int64_t>(256, (grand + 63) / 64);
    hipLaunchKernelGGL(mlp_grad_reduce_onepass_f32, dim3(rb), dim3(256), 0,
                       stream, ra);
    HIP_OK(hipGetLastError());
    std::vector<torch::Tensor> out;
    out.push_back(dx);
    for (int l = 0; l < L; ++l) out.push_back(dws[l]);
    for (int l = 0; l < L; ++l) out.push_back(dbs[l]);
    return out;
  }

  torch::Tensor ws = torch::empty({(int64_t)n_blocks, grand}, opts);
  float* ws_ptr = ws.data_ptr<float>();

  std::vector<torch::Tensor> dws(L), dbs(L);
  torch::Tensor dy = grad_out.contiguous();
  torch::Tensor dx;
  for (int l = L - 1; l >= 0; --l) {
    const int out_d = (int)weights[l].size(0);
    const int in_d = (int)weights[l].size(1);
    torch::Tensor y = (l == L - 1) ? final_out : hidden[l];
    torch::Tensor xin = (l == 0) ? x : hidden[l - 1];
    dws[l] = torch::empty({out_d, in_d}, opts);
    dbs[l] = torch::empty({out_d}, opts);
    dx = torch::empty({batch, in_d}, opts);
    if (maxw == 256) {
      if (input_grad_only) {
        // dgrad chain only (e.g. actor step through a frozen critic:
        // the critic's weight grads would be discarded)
        launch_mlp_dgrad_wide(dy.data_ptr<float>(), y.data_ptr<float>(),
                              weights[l].data_ptr<float>(),
                              dx.data_ptr<float>(), batch, out_d, in_d,
                              (int)acts[l], stream);
      } else {
        // wide layer: 2D-grid dgrad + wgrad kernels
        launch_mlp_bwd_wide(dy.data_ptr<float>(), y.data_ptr<float>(),
                            xin.data_ptr<float>(), weights[l].data_ptr<float>(),
                            dx.data_ptr<float>(), ws_ptr + (long)layer_off[l] * n_blocks, grand,
                            batch, out_d, in_d, (int)acts[l], stream);
      }
    } else {
      int wmode;
      size_t lds_bytes;
      pick_wstage(weights, rows, maxw, true, l, &wmode, &lds_bytes);
      // merged dgrad + wgrad/bias partials in one kernel
      launch_mlp_bwd_layer(dy.data_ptr<float>(), y.data_ptr<float>(),
                           xin.data_ptr<float>(), weights[l].data_ptr<float>(),
                           dx.data_ptr<float>(), ws_ptr + (long)layer_off[l] * n_blocks, grand,
                           batch, out_d, in_d, (int)acts[l], rows, maxw,
                           n_blocks, wmode, lds_bytes, stream);
    }
    HIP_OK(hipGetLastError());
    dy = dx;
  }

  if (input_grad_only) return {dx};
  if (fuse_adam) {
    launch_reduce_adam(weights, biases, totals, ws_ptr, grand, n_blocks, grand,
                       *adam_m, *adam_v, *adam_step, lr, beta1, beta2, eps,
                       weight_decay, adam_step_delta, adam_gate, stream);
    return {dx};
  }
  // one-pass deterministic reduction over the partial rows
  ReduceAllArgs ra{};
  ra.ws = ws_ptr;
  ra.stride = grand;
  ra.n_layers = L;
  ra.n_blocks = n_blocks;
  for (int l = 0; l < L; ++l) {
    ra.dw[l] = dws[l].data_ptr<float>();
    ra.db[l] = dbs[l].data_ptr<float>();
    ra.total[l] = (int)totals[l];
    ra.wsize[l] = (int)(weights[l].size(0) * weights[l].size(1));
  }
  int rb = (int)std::min<int64_t>(256, (grand + 63) / 64);
  hipLaunchKernelGGL(mlp_grad_reduce_onepass_f32, dim3(rb), dim3(256), 0,
                     stream, ra);
  HIP_OK(hipGetLastError());

  std::vector<torch::Tensor> out;
  out.push_back(dx);
  for (int l = 0; l < L; ++l) out.push_back(dws[l]);
  for (int l = 0; l < L; ++l) out.push_back(dbs[l]);
  return out;
}

// ---------------------------------------------------------------------------
// fused losses (loss_kernels.hip)
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> gaussian_policy_loss(torch::Tensor mean,
                                                torch::Tensor actions,
                                                torch::Tensor old_logp,
                                                torch::Tensor adv,
                                                torch::Tensor log_std,
                                                double clip, int64_t mode) {
  check_f32_gpu(mean, "mean");
  const int B = (int)mean.size(0);
  const int D = (int)mean.size(1);
  TORCH_CHECK(D >= 1 && D <= 512,
              "action dim must be in [1,512] for the fused loss");
  auto opts = mean.options();
  auto dmean = torch::empty_like(mean);
  auto dlog_std = torch::empty({D}, opts);
  auto scalars = torch::empty({1}, opts);
  const int n_blocks = std::min(32, (B + 255) / 256);
  auto partials = torch::empty({n_blocks, D + 1}, opts);
  // generic-D kernel scratch (per-row dlogp coefficients)
  auto c_buf = D > 8 ? torch::empty({B}, opts) : torch::Tensor();
  const float* olp = mode == 1 ? old_logp.data_ptr<float>() : nullptr;
  auto stream = current_stream();
  launch_gaussian_loss(mean.data_ptr<float>(), actions.data_ptr<float>(), olp,
                       adv.data_ptr<float>(), log_std.data_ptr<float>(),
                       dmean.data_ptr<float>(),
                       D > 8 ? c_buf.data_ptr<float>() : nullptr,
                       partials.data_ptr<float>(), B,
                       D, (float)clip, (int)mode, n_blocks, stream);
  HIP_OK(hipGetLastError());
  hipLaunchKernelGGL(loss_partials_finalize, dim3(1), dim3(D + 1), 0, stream,
                     partials.data_ptr<float>(), dlog_std.data_ptr<float>(),
                     scalars.data_ptr<float>(), n_blocks, D);
  HIP_OK(hipGetLastError());
  return {dmean, dlog_std, scalars};
}

torch::Tensor gaussian_logp(torch::Tensor mean, torch::Tensor actions,
                            torch::Tensor log_std) {
  check_f32_gpu(mean, "mean");
  const int B = (int)mean.size(0);
  const int D = (int)mean.size(1);
  auto logp = torch::empty({B}, mean.options());
  hipLaunchKernelGGL(gaussian_logp_kernel, dim3(std::min(64, (B + 1023) / 1024)),
                     dim3(1024), 0, current_stream(), mean.data_ptr<float>(),
                     actions.data_ptr<float>(), log_std.data_ptr<float>(),
                     logp.data_ptr<float>(), B, D);
  HIP_OK(hipGetLastError());
  return logp;
}

torch::Tensor gaussian_kl(torch::Tensor mean, torch::Tensor actions,
                          torch::Tensor log_std, torch::Tensor old_logp) {
  check_f32_gpu(mean, "mean");
  const int B = (int)mean.size(0);
  const int D = (int)mean.size(1);
  auto out = torch::empty({1}, mean.options());
  hipLaunchKernelGGL(gaussian_kl_kernel, dim3(1), dim3(1024), 0, current_stream(),
                     mean.data_ptr<float>(), actions.data_ptr<float>(),
                     log_std.data_ptr<float>(), old_logp.data_ptr<float>(),
                     out.data_ptr<float>(), B, D);
  HIP_OK(hipGetLastError());
  return out;
}

std::vector<torch::Tensor> categorical_policy_loss(torch::Tensor logits,
                                                   torch::Tensor actions,
                                                   torch::Tensor old_logp,
                                                   torch::Tensor adv, double clip,
                                                   int64_t mode) {
  check_f32_gpu(logits, "logits");
  const int B = (int)logits.size(0);
  const int N = (int)logits.size(1);
  auto dlogits = torch::empty_like(logits);
  auto scalars = torch::empty({1}, logits.options());
  const int n_blocks = std::min(32, (B + 255) / 256);
  auto partials = torch::empty({n_blocks}, logits.options());
  const float* olp = mode == 1 ? old_logp.data_ptr<float>() : nullptr;
  auto stream = current_stream();
  hipLaunchKernelGGL(categorical_policy_loss_bwd, dim3(n_blocks), dim3(256), 0,
                     stream, logits.data_ptr<float>(),
                     actions.data_ptr<float>(), olp, adv.data_ptr<float>(),
                     dlogits.data_ptr<float>(), partials.data_ptr<float>(), B, N,
                     (float)clip, (int)mode);
  HIP_OK(hipGetLastError());
  hipLaunchKernelGGL(loss_partials_finalize, dim3(1), dim3(1), 0, stream,
                     partials.data_ptr<float>(), nullptr,
                     scalars.data_ptr<float>(), n_blocks, 0);
  HIP_OK(hipGetLastError());
  return {dlogits, scalars};
}

torch::Tensor categorical_logp(torch::Tensor logits, torch::Tensor actions) {
  check_f32_gpu(logits, "logits");
  const int B = (int)logits.size(0);
  const int N = (int)logits.size(1);
  auto logp = torch::empty({B}, logits.options());
  hipLaunchKernelGGL(categorical_logp_kernel, dim3(std::min(64, (B + 1023) / 1024)),
                     dim3(1024), 0, current_stream(), logits.data_ptr<float>(),
                     actions.data_ptr<float>(), logp.data_ptr<float>(), B, N);
  HIP_OK(hipGetLastError());
  return logp;
}

torch::Tensor categorical_kl(torch::Tensor logits, torch::Tensor actions,
                             torch::Tensor old_logp) {
  check_f32_gpu(logits, "logits");
  const int B = (int)logits.size(0);
  const int N = (int)logits.size(1);
  auto out = torch::empty({1}, logits.options());
  hipLaunchKernelGGL(categorical_kl_kernel, dim3(1), dim3(1024), 0,
                     current_stream(), logits.data_ptr<float>(),
                     actions.data_ptr<float>(), old_logp.data_ptr<float>(),
                     out.data_ptr<float>(), B, N);
  HIP_OK(hipGetLastError());
  return out;
}

std::vector<torch::Tensor> value_mse_loss(torch::Tensor v, torch::Tensor ret) {
  check_f32_gpu(v, "v");
  const int B = (int)v.numel();
  auto dv = torch::empty_like(v);
  auto scalars = torch::empty({1}, v.options());
  hipLaunchKernelGGL(value_mse_bwd_kernel, dim3(1), dim3(1024), 0,
                     current_stream(), v.data_ptr<float>(), ret.data_ptr<float>(),
                     dv.data_ptr<float>(), scalars.data_ptr<float>(), B);
  HIP_OK(hipGetLastError());
  return {dv, scalars};
}

// value-function backward with the MSE loss fused into the dZ seed:
// returns [dx, dW..., db..., loss_scalar].  Requires a narrow net with
// identity output (the ValueFunction case); falls back is the caller's
// job (value_supported gates on MLP shape).
std::vector<torch::Tensor> value_mlp_backward(torch::Tensor x,
                                              std::vector<torch::Tensor> weights,
                                              std::vector<torch::Tensor> biases,
                                              std::vector<torch::Tensor> hidden,
                                              torch::Tensor final_out,
                                              std::vector<int64_t> acts,
                                              torch::Tensor returns,
                                              int64_t compute_bf16,
                                              c10::optional<torch::Tensor> partials_out,
                                              c10::optional<std::vector<torch::Tensor>> adam_m,
                                              c10::optional<std::vector<torch::Tensor>> adam_v,
                                              c10::optional<torch::Tensor> adam_step,
                                              double lr, double beta1,
                                              double beta2, double eps,
                                              double weight_decay,
                                              double adam_step_delta,
                                              bool fwd_in_kernel) {
  const bool fuse_adam = adam_m.has_value();
  TORCH_CHECK(!fwd_in_kernel || compute_bf16 == 0,
              "fwd_in_kernel is fp32-only");
  const int L = (int)weights.size();
  check_f32_gpu(x, "x");
  check_f32_gpu(returns, "returns");
  TORCH_CHECK((int)weights[L - 1].size(0) == 1 && acts[L - 1] == 0,
              "value_mlp_backward needs a 1-output identity head");
  const int batch = (int)x.size(0);
  int max_width = (int)x.size(1);
  for (int l = 0; l < L; ++l)
    max_width = std::max(max_width, (int)weights[l].size(0));
  TORCH_CHECK(max_width <= 64, "value_mlp_backward supports narrow nets only");
  auto opts = x.options();
  auto stream = current_stream();

  std::vector<int64_t> totals(L), layer_off(L);
  int64_t grand = 0;
  for (int l = 0; l < L; ++l) {
    totals[l] = (int64_t)weights[l].size(0) * weights[l].size(1) + weights[l].size(0);
    layer_off[l] = grand;
    grand += totals[l];
  }
  size_t whole_w = 0;
  for (auto& w : weights) whole_w += (size_t)w.size(0) * (w.size(1) + 1);
  const int brows = bwd_fused_rows(batch);
  const size_t act_tiles = (size_t)(fwd_in_kernel ? 3 + L : 3);
  const size_t fused_lds = (act_tiles * brows * 68 + whole_w) * 4;
  TORCH_CHECK(fused_lds <= 100 * 1024, "net too large for fused value backward");

  const int fb = (batch + brows - 1) / brows;
  torch::Tensor ws = torch::empty({(int64_t)fb, grand}, opts);
  // partials_out: caller-owned slice -> finalize is deferred (the captured
  // value loop batches all iterations' finalizes into one kernel)
  const bool deferred = partials_out.has_value();
  if (deferred)
    TORCH_CHECK(partials_out->numel() >= fb && partials_out->is_contiguous(),
                "partials_out too small");
  torch::Tensor loss_partials = deferred ? *partials_out : torch::empty({fb}, opts);
  torch::Tensor scalars = torch::empty({1}, opts);
  std::vector<torch::Tensor> dws(L), dbs(L);
  MLPBwdArgs ba{};
  ba.n_layers = L;
  ba.batch = batch;
  ba.ws_stride = grand;
  ba.dims[0] = (int)x.size(1);
  for (int l = 0; l < L; ++l) {
    ba.w[l] = weights[l].data_ptr<float>();
    // fwd_in_kernel: activations never exist in HBM — the kernel
    // computes them into LDS (hidden/final_out are ignored)
    ba.h[l] = fwd_in_kernel
                  ? nullptr
                  : (l == L - 1 ? final_out : hidden[l]).data_ptr<float>();
    ba.b[l] = biases[l].data_ptr<float>();
    ba.dims[l + 1] = (int)weights[l].size(0);
    ba.acts[l] = (int)acts[l];
    ba.layer_off[l] = (int)layer_off[l];
    dws[l] = torch::empty({(int)weights[l].size(0), (int)weights[l].size(1)}, opts);
    dbs[l] = torch::empty({(int)weights[l].size(0)}, opts);
  }
  torch::Tensor dx = torch::empty({batch, x.size(1)}, opts);
  launch_mlp_bwd_fused(ba, x.data_ptr<float>(), nullptr, dx.data_ptr<float>(),
                       ws.data_ptr<float>(), returns.data_ptr<float>(),
                       loss_partials.data_ptr<float>(), fused_lds, fb,
                       (int)compute_bf16, stream, brows, fwd_in_kernel ? 1 : 0);
  HIP_OK(hipGetLastError());

  if (fuse_adam) {
    launch_reduce_adam(weights, biases, totals, ws.data_ptr<float>(), grand,
                       fb, grand, *adam_m, *adam_v, *adam_step, lr, beta1,
                       beta2, eps, weight_decay, adam_step_delta,
                       c10::nullopt, stream);
  } else {
    ReduceAllArgs ra{};
    ra.ws = ws.data_ptr<float>();
    ra.stride = grand;
    ra.n_layers = L;
    ra.n_blocks = fb;
    for (int l = 0; l < L; ++l) {
      ra.dw[l] = dws[l].data_ptr<float>();
      ra.db[l] = dbs[l].data_ptr<float>();
      ra.total[l] = (int)totals[l];
      ra.wsize[l] = (int)(weights[l].size(0) * weights[l].size(1));
    }
    int rb = (int)std::min<int64_t>(256, (grand + 63) / 64);
    hipLaunchKernelGGL(mlp_grad_reduce_onepass_f32, dim3(rb), dim3(256), 0,
                       stream, ra);
    HIP_OK(hipGetLastError());
  }
  if (!deferred) {
    // loss = sum of the per-block partials (fixed order)
    hipLaunchKernelGGL(loss_partials_finalize, dim3(1), dim3(1), 0, stream,
                       loss_partials.data_ptr<float>(), nullptr,
                       scalars.data_ptr<float>(), fb, 0);
    HIP_OK(hipGetLastError());
  }

  std::vector<torch::Tensor> out;
  out.push_back(dx);
  for (int l = 0; l < L; ++l) out.push_back(dws[l]);
  for (int l = 0; l < L; ++l) out.push_back(dbs[l]);
  out.push_back(scalars);
  return out;
}

// ---------------------------------------------------------------------------
// segmented GAE
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> segmented_gae(torch::Tensor rewards, torch::Tensor values,
                                         torch::Tensor last_values,
                                         torch::Tensor offsets, torch::Tensor dones,
                                         double gamma, double lam) {
  check_f32_gpu(rewards, "rewards");
  check_f32_gpu(values, "values");
  check_f32_gpu(last_values, "last_values");
  TORCH_CHECK(offsets.scalar_type() == torch::kInt32 && offsets.is_cuda());
  TORCH_CHECK(dones.scalar_type() == torch::kInt32 && dones.is_cuda());
  const int n_eps = (int)last_values.size(0);
  auto adv = torch::empty_like(rewards);
  auto ret = torch::empty_like(rewards);
  if (n_eps > 0) {
    hipLaunchKernelGGL(segmented_gae_kernel, dim3(n_eps), dim3(256), 0,
                       current_stream(), rewards.data_ptr<float>(),
                       values.data_ptr<float>(), last_values.data_ptr<float>(),
                       offsets.data_ptr<int>(), dones.data_ptr<int>(),
                       adv.data_ptr<float>(), ret.data_ptr<float>(), (float)gamma,
                       (float)lam);
    HIP_OK(hipGetLastError());
  }
  return {adv, ret};
}

torch::Tensor normalize(torch::Tensor x) {
  check_f32_gpu(x, "x");
  auto y = torch::empty_like(x);
  hipLaunchKernelGGL(normalize_kernel, dim3(1), dim3(1024), 0, current_stream(),
                     x.data_ptr<float>(), y.data_ptr<float>(), (int)x.numel());
  HIP_OK(hipGetLastError());
  return y;
}

torch::Tensor q_target(torch::Tensor r, torch::Tensor d, torch::Tensor qn,
                       double gamma) {
  check_f32_gpu(r, "rewards");
  auto out = torch::empty_like(r);
  int n = (int)r.numel();
  hipLaunchKernelGGL(q_target_kernel, dim3((n + 255) / 256), dim3(256), 0,
                     current_stream(), r.data_ptr<float>(), d.data_ptr<float>(),
                     qn.data_ptr<float>(), out.data_ptr<float>(), (float)gamma, n);
  HIP_OK(hipGetLastError());
  return out;
}

// ---------------------------------------------------------------------------
// fused updates
// ---------------------------------------------------------------------------
void fused_adam_(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs,
                 std::vector<torch::Tensor> exp_avg_sqs,
                 std::vector<torch::Tensor> steps, double lr, double beta1,
                 double beta2, double eps, double weight_decay,
                 double step_delta, bool do_bump,
                 c10::optional<torch::Tensor> gate) {
  const float* gate_ptr = nullptr;
  if (gate.has_value()) {
    TORCH_CHECK(gate->scalar_type() == torch::kFloat32 && gate->is_cuda() &&
                    gate->numel() == 1,
                "gate must be a 1-element fp32 CUDA tensor");
    gate_ptr = gate->data_ptr<float>();
  }
  size_t n = params.size();
  for (size_t base = 0; base < n; base += MT_MAX_TENSORS) {
    AdamArgs a{};
    a.n_tensors = (int)std::min((size_t)MT_MAX_TENSORS, n - base);
    for (int i = 0; i < a.n_tensors; ++i) {
      size_t t = base + i;
      a.p[i] = params[t].data_ptr<float>();
      a.g[i] = grads[t].data_ptr<float>();
      a.m[i] = exp_avgs[t].data_ptr<float>();
      a.v[i] = exp_avg_sqs[t].data_ptr<float>();
      a.step[i] = steps[t].data_ptr<float>();
      a.numel[i] = (int)params[t].numel();
    }
    a.lr = (float)lr;
    a.beta1 = (float)beta1;
    a.beta2 = (float)beta2;
    a.eps = (float)eps;
    a.weight_decay = (float)weight_decay;
    a.step_delta = (float)step_delta;
    a.gate = gate_ptr;
    int max_chunks = 1;
    for (int i = 0; i < a.n_tensors; ++i)
      max_chunks = std::max(max_chunks, (a.numel[i] + 8191) / 8192);
    hipLaunchKernelGGL(fused_adam_kernel, dim3(a.n_tensors, max_chunks),
                       dim3(256), 0, current_stream(), a);
    HIP_OK(hipGetLastError());
    if (do_bump) {
      a.step_delta = 1.f;
      hipLaunchKernelGGL(adam_step_bump_kernel, dim3(1), dim3(MT_MAX_TENSORS), 0,
                         current_stream(), a);
      HIP_OK(hipGetLastError());
    }
  }
}

__global__ void adam_step_bump_dev_kernel(AdamArgs a, const float* amount);

// bump by a device scalar (captured gated policy loop)
void adam_bump_dev_(std::vector<torch::Tensor> steps, torch::Tensor amount) {
  size_t n = steps.size();
  for (size_t base = 0; base < n; base += MT_MAX_TENSORS) {
    AdamArgs a{};
    a.n_tensors = (int)std::min((size_t)MT_MAX_TENSORS, n - base);
    for (int i = 0; i < a.n_tensors; ++i)
      a.step[i] = steps[base + i].data_ptr<float>();
    hipLaunchKernelGGL(adam_step_bump_dev_kernel, dim3(1), dim3(MT_MAX_TENSORS),
                       0, current_stream(), a, amount.data_ptr<float>());
    HIP_OK(hipGetLastError());
  }
}

// bump-only entry (captured value loop: ONE bump of +num_iters per replay)
void adam_bump_(std::vector<torch::Tensor> steps, double amount) {
  size_t n = steps.size();
  for (size_t base = 0; base < n; base += MT_MAX_TENSORS) {
    AdamArgs a{};
    a.n_tensors = (int)std::min((size_t)MT_MAX_TENSORS, n - base);
    for (int i = 0; i < a.n_tensors; ++i)
      a.step[i] = steps[base + i].data_ptr<float>();
    a.step_delta = (float)amount;
    hipLaunchKernelGGL(adam_step_bump_kernel, dim3(1), dim3(MT_MAX_TENSORS), 0,
                       current_stream(), a);
    HIP_OK(hipGetLastError());
  }
}

// deferred value-loss finalize over all iterations at once
torch::Tensor value_loss_finalize(torch::Tensor partials, int64_t fb) {
  check_f32_gpu(partials, "partials");
  const int rows = (int)partials.size(0);
  const int stride = (int)partials.size(1);
  TORCH_CHECK((int)fb <= stride, "fb exceeds partials row stride");
  auto scalars = torch::empty({rows}, partials.options());
  hipLaunchKernelGGL(value_loss_finalize_rows, dim3((rows + 255) / 256),
                     dim3(256), 0, current_stream(),
                     partials.data_ptr<float>(), scalars.data_ptr<float>(),
                     rows, (int)fb, stride);
  HIP_OK(hipGetLastError());
  return scalars;
}

void fused_polyak_(std::vector<torch::Tensor> srcs, std::vector<torch::Tensor> dsts,
                   double rho) {
  size_t n = srcs.size();
  for (size_t base = 0; base < n; base += MT_MAX_TENSORS) {
    PolyakArgs a{};
    a.n_tensors = (int)std::min((size_t)MT_MAX_TENSORS, n - base);
    for (int i = 0; i < a.n_tensors; ++i) {
      size_t t = base + i;
      a.src[i] = srcs[t].data_ptr<float>();
      a.dst[i] = dsts[t].data_ptr<float>();
      a.numel[i] = (int)srcs[t].numel();
    }
    a.rho = (float)rho;
    int max_chunks = 1;
    for (int i = 0; i < a.n_tensors; ++i)
      max_chunks = std::max(max_chunks, (a.numel[i] + 8191) / 8192);
    hipLaunchKernelGGL(fused_polyak_kernel, dim3(a.n_tensors, max_chunks),
                       dim3(256), 0, current_stream(), a);
    HIP_OK(hipGetLastError());
  }
}

static const unsigned long long* ctr_ptr(const c10::optional<torch::Tensor>& t) {
  if (!t.has_value()) return nullptr;
  TORCH_CHECK(t->scalar_type() == torch::kInt64 && t->is_cuda() && t->numel() == 1,
              "offset_ctr must be a 1-element int64 CUDA tensor");
  return reinterpret_cast<const unsigned long long*>(t->data_ptr<int64_t>());
}

torch::Tensor gaussian_sample(torch::Tensor mean, torch::Tensor log_std,
                              int64_t seed, int64_t offset, double noise_scale,
                              double limit,
                              c10::optional<torch::Tensor> offset_ctr,
                              c10::optional<torch::Tensor> out_opt) {
  check_f32_gpu(mean, "mean");
  const int B = (int)mean.size(0);
  const int D = (int)mean.size(1);
  auto out = out_opt.has_value() ? *out_opt : torch::empty_like(mean);
  const int total = B * D;
  hipLaunchKernelGGL(gaussian_sample_kernel,
                     dim3(std::min(256, (total + 255) / 256)), dim3(256), 0,
                     current_stream(), mean.data_ptr<float>(),
                     log_std.data_ptr<float>(), out.data_ptr<float>(), B, D,
                     (uint64_t)seed, (uint64_t)offset, (float)noise_scale,
                     (float)limit, ctr_ptr(offset_ctr));
  HIP_OK(hipGetLastError());
  return out;
}

torch::Tensor categorical_sample(torch::Tensor logits, int64_t seed,
                                 int64_t offset) {
  check_f32_gpu(logits, "logits");
  const int B = (int)logits.size(0);
  const int N = (int)logits.size(1);
  auto out = torch::empty({B}, logits.options().dtype(torch::kInt64));
  hipLaunchKernelGGL(categorical_sample_kernel,
                     dim3(std::min(256, (B + 255) / 256)), dim3(256), 0,
                     current_stream(), logits.data_ptr<float>(),
                     out.data_ptr<int64_t>(), B, N, (uint64_t)seed,
                     (uint64_t)offset);
  HIP_OK(hipGetLastError());
  return out;
}

std::vector<torch::Tensor> synthetic_env_step(torch::Tensor state,
                                              torch::Tensor actions,
                                              torch::Tensor A, torch::Tensor B,
                                              torch::Tensor w, double sigma,
                                              int64_t seed, int64_t offset,
                                              bool do_reset,
                                              c10::optional<torch::Tensor> offset_ctr,
                                              c10::optional<torch::Tensor> s_out_opt,
                                              c10::optional<torch::Tensor> final_out_opt,
                                              c10::optional<torch::Tensor> reward_opt) {
  check_f32_gpu(state, "state");
  check_f32_gpu(actions, "actions");
  const int N = (int)state.size(0);
  const int O = (int)state.size(1);
  const int Adim = (int)actions.size(1);
  TORCH_CHECK(O <= 64, "synthetic_env_step: obs_dim must be <= 64 (one wave per row)");
  auto s_out = s_out_opt.has_value() ? *s_out_opt : torch::empty_like(state);
  auto final_out = final_out_opt.has_value() ? *final_out_opt
                   : (do_reset ? torch::empty_like(state) : s_out);
  auto reward = reward_opt.has_value() ? *reward_opt
                                       : torch::empty({N}, state.options());
  hipLaunchKernelGGL(synthetic_env_step_kernel, dim3(N), dim3(64), 0,
                     current_stream(), state.data_ptr<float>(),
                     actions.data_ptr<float>(), A.data_ptr<float>(),
                     B.data_ptr<float>(), w.data_ptr<float>(),
                     s_out.data_ptr<float>(), final_out.data_ptr<float>(),
                     reward.data_ptr<float>(), N, O, Adim, (float)sigma,
                     (uint64_t)seed, (uint64_t)offset, do_reset ? 1 : 0,
                     ctr_ptr(offset_ctr));
  HIP_OK(hipGetLastError());
  return {s_out, reward, final_out};
}

torch::Tensor synthetic_env_reset(int64_t num_envs, int64_t obs_dim,
                                  torch::Tensor like, int64_t seed,
                                  int64_t offset,
                                  c10::optional<torch::Tensor> offset_ctr,
                                  c10::optional<torch::Tensor> out_opt) {
  auto s_out = out_opt.has_value()
                   ? *out_opt
                   : torch::empty({num_envs, obs_dim}, like.options());
  const int total = (int)(num_envs * obs_dim);
  hipLaunchKernelGGL(synthetic_env_reset_kernel,
                     dim3(std::min(256, (total + 255) / 256)), dim3(256), 0,
                     current_stream(), s_out.data_ptr<float>(), total,
                     (uint64_t)seed, (uint64_t)offset, ctr_ptr(offset_ctr));
  HIP_OK(hipGetLastError());
  return s_out;
}

torch::Tensor td3_smooth(torch::Tensor a, int64_t seed, int64_t offset,
                         double scale, double clip, double limit,
                         c10::optional<torch::Tensor> offset_ctr) {
  check_f32_gpu(a, "actions");
  auto out = torch::empty_like(a);
  const int total = (int)a.numel();
  hipLaunchKernelGGL(td3_smooth_kernel,
                     dim3(std::min(256, (total + 255) / 256)), dim3(256), 0,
                     current_stream(), a.data_ptr<float>(),
                     out.data_ptr<float>(), total, (uint64_t)seed,
                     (uint64_t)offset, (float)scale, (float)clip,
                     (float)limit, ctr_ptr(offset_ctr));
  HIP_OK(hipGetLastError());
  return out;
}

torch::Tensor q_target_min2(torch::Tensor r, torch::Tensor d, torch::Tensor q1,
                            torch::Tensor q2, double gamma) {
  check_f32_gpu(r, "rewards");
  auto out = torch::empty_like(r);
  const int n = (int)r.numel();
  hipLaunchKernelGGL(q_target_min2_kernel, dim3((n + 255) / 256), dim3(256), 0,
                     current_stream(), r.data_ptr<float>(), d.data_ptr<float>(),
                     q1.data_ptr<float>(), q2.data_ptr<float>(),
                     out.data_ptr<float>(), (float)gamma, n);
  HIP_OK(hipGetLastError());
  return out;
}

std::vector<torch::Tensor> replay_gather(torch::Tensor obs, torch::Tensor act,
                                         torch::Tensor rew, torch::Tensor nxt,
                                         torch::Tensor dn, torch::Tensor size_dev,
                                         int64_t B, int64_t seed, int64_t offset,
                                         c10::optional<torch::Tensor> offset_ctr) {
  check_f32_gpu(obs, "obs");
  check_f32_gpu(act, "act");
  check_f32_gpu(rew, "rew");
  check_f32_gpu(nxt, "nxt");
  check_f32_gpu(dn, "dn");
  TORCH_CHECK(size_dev.scalar_type() == torch::kInt64 && size_dev.is_cuda() &&
                  size_dev.numel() == 1,
              "size_dev must be a 1-element int64 CUDA tensor");
  ReplayGatherArgs a{};
  a.obs = obs.data_ptr<float>();
  a.act = act.data_ptr<float>();
  a.rew = rew.data_ptr<float>();
  a.nxt = nxt.data_ptr<float>();
  a.dn = dn.data_ptr<float>();
  a.size = reinterpret_cast<const long long*>(size_dev.data_ptr<int64_t>());
  a.B = (int)B;
  a.O = (int)obs.size(1);
  a.A = (int)act.size(1);
  auto qin = torch::empty({B, a.O + a.A}, obs.options());
  auto obs_out = torch::empty({B, a.O}, obs.options());
  auto nxt_out = torch::empty({B, a.O}, obs.options());
  auto rew_out = torch::empty({B}, obs.options());
  auto dn_out = torch::empty({B}, obs.options());
  a.qin = qin.data_ptr<float>();
  a.obs_out = obs_out.data_ptr<float>();
  a.nxt_out = nxt_out.data_ptr<float>();
  a.rew_out = rew_out.data_ptr<float>();
  a.dn_out = dn_out.data_ptr<float>();
  a.seed = (uint64_t)seed;
  a.offset = (uint64_t)offset;
  a.offset_ptr = ctr_ptr(offset_ctr);
  const long total = (long)a.B * (2 * a.O + a.A + 2);
  hipLaunchKernelGGL(replay_gather_kernel,
                     dim3((int)std::min<long>(512, (total + 255) / 256)),
                     dim3(256), 0, current_stream(), a);
  HIP_OK(hipGetLastError());
  return {qin, obs_out, nxt_out, rew_out, dn_out};
}

void ppo_gate_update_(torch::Tensor gate, torch::Tensor kl,
                      torch::Tensor kl_final, torch::Tensor iters_done,
                      double thr) {
  hipLaunchKernelGGL(ppo_gate_update_kernel, dim3(1), dim3(1), 0,
                     current_stream(), gate.data_ptr<float>(),
                     kl.data_ptr<float>(), kl_final.data_ptr<float>(),
                     iters_done.data_ptr<float>(), (float)thr);
  HIP_OK(hipGetLastError());
}

// ONE captured Gaussian-PPO policy iteration = 3 kernels:
//   1. DO_FWD fused kernel: forward (LDS activations) + pending-KL /
//      loss / dlog_std partials + clipped-surrogate dZ seed + whole-net
//      backward stage-1
//   2. gate reduce: close the gate if the pending KL crossed 1.5*maxkl
//      (so this iteration's update is skipped, like the reference's
//      break at ppo.py:176-181)
//   3. merged reduce+Adam over net params AND log_std (pseudo-layer),
//      gated
// Narrow fp32 identity-head Gaussian policies only (the bench config).
torch::Tensor gaussian_ppo_policy_iter(
    torch::Tensor x, std::vector<torch::Tensor> weights,
    std::vector<torch::Tensor> biases, std::vector<int64_t> acts,
    torch::Tensor actions, torch::Tensor old_logp, torch::Tensor advantages,
    torch::Tensor log_std, double clip, std::vector<torch::Tensor> adam_m,
    std::vector<torch::Tensor> adam_v, torch::Tensor ls_m, torch::Tensor ls_v,
    torch::Tensor adam_step, double lr, double beta1, double beta2, double eps,
    double weight_decay, double step_delta, torch::Tensor gate,
    torch::Tensor kl_final, torch::Tensor iters_done, double thr,
    torch::Tensor loss_out, bool first_iter) {
  const int L = (int)weights.size();
  check_f32_gpu(x, "x");
  const int batch = (int)x.size(0);
  const int D = (int)weights[L - 1].size(0);
  TORCH_CHECK(acts[L - 1] == 0, "needs an identity head");
  TORCH_CHECK(L + 1 <= MLP_MAX_LAYERS, "too many layers for the log_std slot");
  TORCH_CHECK(D <= 64, "action dim must fit one LDS row");
  int max_width = (int)x.size(1);
  for (int l = 0; l < L; ++l)
    max_width = std::max(max_width, (int)weights[l].size(0));
  TORCH_CHECK(max_width <= 64, "narrow nets only");
  auto opts = x.options();
  auto stream = current_stream();

  std::vector<int64_t> totals(L), layer_off(L);
  int64_t grand = 0;
  for (int l = 0; l < L; ++l) {
    totals[l] = (int64_t)weights[l].size(0) * weights[l].size(1) + weights[l].size(0);
    layer_off[l] = grand;
    grand += totals[l];
  }
  const int64_t grand_ls = grand + D;  // + dlog_std pseudo-layer
  size_t whole_w = 0;
  for (auto& w : weights) whole_w += (size_t)w.size(0) * (w.size(1) + 1);
  const int brows = bwd_fused_rows(batch);
  const size_t fused_lds = (((size_t)3 + L) * brows * 68 + whole_w) * 4;
  TORCH_CHECK(fused_lds <= 100 * 1024, "net too large for the fused iter");
  const int fb = (batch + brows - 1) / brows;

  torch::Tensor ws = torch::empty({(int64_t)fb, grand_ls}, opts);
  torch::Tensor loss_partials = torch::empty({fb}, opts);
  torch::Tensor kl_partials = torch::empty({fb}, opts);
  torch::Tensor dx = torch::empty({batch, x.size(1)}, opts);

  MLPBwdArgs ba{};
  ba.n_layers = L;
  ba.batch = batch;
  ba.ws_stride = grand_ls;
  ba.dims[0] = (int)x.size(1);
  for (int l = 0; l < L; ++l) {
    ba.w[l] = weights[l].data_ptr<float>();
    ba.h[l] = nullptr;
    ba.b[l] = biases[l].data_ptr<float>();
    ba.dims[l + 1] = (int)weights[l].size(0);
    ba.acts[l] = (int)acts[l];
    ba.layer_off[l] = (int)layer_off[l];
  }
  GaussSeedArgs ga{};
  ga.actions = actions.data_ptr<float>();
  ga.old_logp = old_logp.data_ptr<float>();
  ga.adv = advantages.data_ptr<float>();
  ga.log_std = log_std.data_ptr<float>();
  ga.kl_partials = kl_partials.data_ptr<float>();
  ga.dls_off = (int)grand;
  ga.clip = (float)clip;

  launch_mlp_bwd_fused(ba, x.data_ptr<float>(), nullptr, dx.data_ptr<float>(),
                       ws.data_ptr<float>(), nullptr,
                       loss_partials.data_ptr<float>(), fused_lds, fb, 0,
                       stream, brows, 1, ga);
  HIP_OK(hipGetLastError());

  hipLaunchKernelGGL(ppo_gate_update_reduce_kernel, dim3(1), dim3(1), 0,
                     stream, gate.data_ptr<float>(),
                     kl_partials.data_ptr<float>(),
                     loss_partials.data_ptr<float>(), fb, 1.f / (float)batch,
                     kl_final.data_ptr<float>(), iters_done.data_ptr<float>(),
                     (float)thr, loss_out.data_ptr<float>(),
                     first_iter ? 1 : 0);
  HIP_OK(hipGetLastError());

  TORCH_CHECK((int)adam_m.size() == 2 * L && (int)adam_v.size() == 2 * L);
  ReduceAdamArgs ra{};
  ra.ws = ws.data_ptr<float>();
  ra.stride = grand_ls;
  ra.n_layers = L + 1;
  ra.n_blocks = fb;
  for (int l = 0; l < L; ++l) {
    ra.pw[l] = weights[l].data_ptr<float>();
    ra.pb[l] = biases[l].data_ptr<float>();
    ra.mw[l] = adam_m[l].data_ptr<float>();
    ra.mb[l] = adam_m[L + l].data_ptr<float>();
    ra.vw[l] = adam_v[l].data_ptr<float>();
    ra.vb[l] = adam_v[L + l].data_ptr<float>();
    ra.total[l] = (int)totals[l];
    ra.wsize[l] = (int)(weights[l].size(0) * weights[l].size(1));
  }
  // log_std pseudo-layer: bias-only (wsize 0)
  ra.pw[L] = nullptr;
  ra.pb[L] = log_std.data_ptr<float>();
  ra.mw[L] = nullptr;
  ra.mb[L] = ls_m.data_ptr<float>();
  ra.vw[L] = nullptr;
  ra.vb[L] = ls_v.data_ptr<float>();
  ra.total[L] = D;
  ra.wsize[L] = 0;
  ra.step = adam_step.data_ptr<float>();
  ra.lr = (float)lr;
  ra.beta1 = (float)beta1;
  ra.beta2 = (float)beta2;
  ra.eps = (float)eps;
  ra.weight_decay = (float)weight_decay;
  ra.step_delta = (float)step_delta;
  ra.gate = gate.data_ptr<float>();
  int rb = (int)std::min<int64_t>(256, (grand_ls + 63) / 64);
  hipLaunchKernelGGL(mlp_grad_reduce_adam_f32, dim3(rb), dim3(256), 0, stream,
                     ra);
  HIP_OK(hipGetLastError());
  return dx;
}

void counter_add_(torch::Tensor ctr, int64_t delta) {
  TORCH_CHECK(ctr.scalar_type() == torch::kInt64 && ctr.is_cuda() && ctr.numel() == 1,
              "ctr must be a 1-element int64 CUDA tensor");
  hipLaunchKernelGGL(counter_add_kernel, dim3(1), dim3(1), 0, current_stream(),
                     reinterpret_cast<unsigned long long*>(ctr.data_ptr<int64_t>()),
                     (unsigned long long)delta);
  HIP_OK(hipGetLastError());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("mlp_forward", &mlp_forward, "fused MLP forward (gfx950)",
        py::arg("x"), py::arg("weights"), py::arg("biases"), py::arg("acts"),
        py::arg("save_hidden"), py::arg("compute_bf16") = 0);
  m.def("mlp_backward", &mlp_backward, "fused MLP backward (gfx950)",
        py::arg("grad_out"), py::arg("x"), py::arg("weights"), py::arg("biases"),
        py::arg("hidden"), py::arg("final_out"), py::arg("acts"),
        py::arg("compute_bf16") = 0, py::arg("adam_m") = py::none(),
        py::arg("adam_v") = py::none(), py::arg("adam_step") = py::none(),
        py::arg("lr") = 0.0, py::arg("beta1") = 0.9, py::arg("beta2") = 0.999,
        py::arg("eps") = 1e-8, py::arg("weight_decay") = 0.0,
        py::arg("adam_step_delta") = 0.0, py::arg("adam_gate") = py::none(),
        py::arg("input_grad_only") = false);
  m.def("value_mlp_backward", &value_mlp_backward,
        "fused value-MSE whole-net backward (gfx950)", py::arg("x"),
        py::arg("weights"), py::arg("biases"), py::arg("hidden"),
        py::arg("final_out"), py::arg("acts"), py::arg("returns"),
        py::arg("compute_bf16") = 0, py::arg("partials_out") = py::none(),
        py::arg("adam_m") = py::none(), py::arg("adam_v") = py::none(),
        py::arg("adam_step") = py::none(), py::arg("lr") = 0.0,
        py::arg("beta1") = 0.9, py::arg("beta2") = 0.999,
        py::arg("eps") = 1e-8, py::arg("weight_decay") = 0.0,
        py::arg("adam_step_delta") = 0.0, py::arg("fwd_in_kernel") = false);
  m.def("value_loss_partials_blocks",
        [](int64_t batch) {
          const int brows = bwd_fused_rows((int)batch);
          return (batch + brows - 1) / brows;
        },
        "partials row length used by value_mlp_backward");
  m.def("value_loss_finalize", &value_loss_finalize,
        "batched deferred value-loss finalize (gfx950)");
  m.def("adam_bump_", &adam_bump_, "bump Adam step counters (gfx950)");
  m.def("adam_bump_dev_", &adam_bump_dev_,
        "bump Adam step counters by a device scalar (gfx950)");
  m.def("ppo_gate_update_", &ppo_gate_update_,
        "device-side KL early-stop gate bookkeeping (gfx950)");
  m.def("gaussian_ppo_policy_iter", &gaussian_ppo_policy_iter,
        "one 3-kernel Gaussian-PPO policy iteration (gfx950)");
  m.def("segmented_gae", &segmented_gae, "segmented GAE+returns scan (gfx950)");
  m.def("normalize", &normalize, "fused mean/std normalize (gfx950)");
  m.def("q_target", &q_target, "fused Q-learning target (gfx950)");
  m.def("td3_smooth", &td3_smooth,
        "TD3 target-policy smoothing: Philox noise + clip + limit (gfx950)",
        py::arg("a"), py::arg("seed"), py::arg("offset"), py::arg("scale"),
        py::arg("clip"), py::arg("limit"), py::arg("offset_ctr") = py::none());
  m.def("q_target_min2", &q_target_min2,
        "fused min-twin Q-learning target (gfx950)");
  m.def("replay_gather", &replay_gather,
        "one-kernel Philox minibatch gather from the HBM replay ring (gfx950)",
        py::arg("obs"), py::arg("act"), py::arg("rew"), py::arg("nxt"),
        py::arg("dn"), py::arg("size_dev"), py::arg("B"), py::arg("seed"),
        py::arg("offset"), py::arg("offset_ctr") = py::none());
  m.def("fused_adam_", &fused_adam_, "fused multi-tensor Adam (gfx950)",
        py::arg("params"), py::arg("grads"), py::arg("exp_avgs"),
        py::arg("exp_avg_sqs"), py::arg("steps"), py::arg("lr"),
        py::arg("beta1"), py::arg("beta2"), py::arg("eps"),
        py::arg("weight_decay"), py::arg("step_delta") = 0.0,
        py::arg("do_bump") = true, py::arg("gate") = py::none());
  m.def("fused_polyak_", &fused_polyak_, "fused multi-tensor Polyak (gfx950)");
  m.def("gaussian_policy_loss", &gaussian_policy_loss,
        "fused Gaussian VPG/PPO loss fwd+bwd (gfx950)");
  m.def("gaussian_logp", &gaussian_logp, "Gaussian log-prob (gfx950)");
  m.def("gaussian_kl", &gaussian_kl, "approx KL for Gaussian policy (gfx950)");
  m.def("categorical_policy_loss", &categorical_policy_loss,
        "fused Categorical VPG/PPO loss fwd+bwd (gfx950)");
  m.def("categorical_logp", &categorical_logp, "Categorical log-prob (gfx950)");
  m.def("categorical_kl", &categorical_kl, "approx KL for Categorical policy (gfx950)");
  m.def("value_mse_loss", &value_mse_loss, "fused value MSE fwd+bwd (gfx950)");
  m.def("gaussian_sample", &gaussian_sample,
        "Philox Gaussian action sample (gfx950)", py::arg("mean"),
        py::arg("log_std"), py::arg("seed"), py::arg("offset"),
        py::arg("noise_scale"), py::arg("limit"),
        py::arg("offset_ctr") = py::none(), py::arg("out") = py::none());
  m.def("categorical_sample", &categorical_sample,
        "Philox categorical action sample (gfx950)");
  m.def("synthetic_env_step", &synthetic_env_step,
        "fused synthetic-env transition + reward (gfx950)", py::arg("state"),
        py::arg("actions"), py::arg("A"), py::arg("B"), py::arg("w"),
        py::arg("sigma"), py::arg("seed"), py::arg("offset"),
        py::arg("do_reset"), py::arg("offset_ctr") = py::none(),
        py::arg("s_out") = py::none(), py::arg("final_out") = py::none(),
        py::arg("reward") = py::none());
  m.def("synthetic_env_reset", &synthetic_env_reset,
        "synthetic-env init states (gfx950)", py::arg("num_envs"),
        py::arg("obs_dim"), py::arg("like"), py::arg("seed"), py::arg("offset"),
        py::arg("offset_ctr") = py::none(), py::arg("out") = py::none());
  m.def("counter_add_", &counter_add_,
        "advance a device RNG counter (gfx950)");
}
