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
__global__ void fused_mlp_fwd_f32(MLPArgs args, const float* x, int save_hidden);
__global__ void mlp_dgrad_f32(const float* dy, const float* y, const float* W,
                              float* dx, int batch, int out_d, int in_d, int act);
__global__ void mlp_wgrad_partial_f32(const float* dy, const float* y,
                                      const float* xin, float* workspace,
                                      int batch, int out_d, int in_d, int act);
__global__ void mlp_grad_reduce_f32(const float* workspace, float* dw, float* db,
                                    int n_blocks, int out_d, int in_d);
__global__ void segmented_gae_kernel(const float* rewards, const float* values,
                                     const float* last_values, const int* offsets,
                                     const int* dones, float* advantages,
                                     float* returns, float gamma, float lam);
__global__ void normalize_kernel(const float* x, float* y, int n);
__global__ void q_target_kernel(const float* r, const float* d, const float* qn,
                                float* out, float gamma, int n);

__global__ void fused_adam_kernel(AdamArgs a);
__global__ void fused_polyak_kernel(PolyakArgs a);

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

}  // namespace

// ---------------------------------------------------------------------------
// fused MLP
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> mlp_forward(torch::Tensor x,
                                       std::vector<torch::Tensor> weights,
                                       std::vector<torch::Tensor> biases,
                                       std::vector<int64_t> acts,
                                       bool save_hidden) {
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

  auto opts = x.options();
  std::vector<torch::Tensor> outs;  // [final, h0..h_{L-2}]
  torch::Tensor final_out = torch::empty({x.size(0), args.dims[L]}, opts);
  outs.push_back(final_out);
  for (int l = 0; l < L - 1; ++l) {
    if (save_hidden) {
      torch::Tensor h = torch::empty({x.size(0), args.dims[l + 1]}, opts);
      args.h[l] = h.data_ptr<float>();
      outs.push_back(h);
    } else {
      args.h[l] = nullptr;
    }
  }
  args.h[L - 1] = final_out.data_ptr<float>();

  const int n_blocks = (args.batch + MLP_ROWS - 1) / MLP_ROWS;
  if (n_blocks > 0) {
    hipLaunchKernelGGL(fused_mlp_fwd_f32, dim3(n_blocks), dim3(256), 0,
                       current_stream(), args, x.data_ptr<float>(),
                       save_hidden ? 1 : 0);
    HIP_OK(hipGetLastError());
  }
  return outs;
}

std::vector<torch::Tensor> mlp_backward(torch::Tensor grad_out, torch::Tensor x,
                                        std::vector<torch::Tensor> weights,
                                        std::vector<torch::Tensor> biases,
                                        std::vector<torch::Tensor> hidden,
                                        torch::Tensor final_out,
                                        std::vector<int64_t> acts) {
  const int L = (int)weights.size();
  check_f32_gpu(grad_out, "grad_out");
  check_f32_gpu(x, "x");
  const int batch = (int)x.size(0);
  const int n_blocks = (batch + MLP_ROWS - 1) / MLP_ROWS;
  auto opts = x.options();
  auto stream = current_stream();

  std::vector<torch::Tensor> dws(L), dbs(L);
  torch::Tensor dy = grad_out.contiguous();
  torch::Tensor dx;
  for (int l = L - 1; l >= 0; --l) {
    const int out_d = (int)weights[l].size(0);
    const int in_d = (int)weights[l].size(1);
    torch::Tensor y = (l == L - 1) ? final_out : hidden[l];
    torch::Tensor xin = (l == 0) ? x : hidden[l - 1];

    // wgrad + bias grad via deterministic split-K workspace
    torch::Tensor ws = torch::empty({(int64_t)n_blocks, (int64_t)out_d * in_d + out_d}, opts);
    hipLaunchKernelGGL(mlp_wgrad_partial_f32, dim3(n_blocks), dim3(256), 0, stream,
                       dy.data_ptr<float>(), y.data_ptr<float>(),
                       xin.data_ptr<float>(), ws.data_ptr<float>(), batch, out_d,
                       in_d, (int)acts[l]);
    HIP_OK(hipGetLastError());
    dws[l] = torch::empty({out_d, in_d}, opts);
    dbs[l] = torch::empty({out_d}, opts);
    int total = out_d * in_d + out_d;
    int rb = std::min(256, (total + 255) / 256);
    hipLaunchKernelGGL(mlp_grad_reduce_f32, dim3(rb), dim3(256), 0, stream,
                       ws.data_ptr<float>(), dws[l].data_ptr<float>(),
                       dbs[l].data_ptr<float>(), n_blocks, out_d, in_d);
    HIP_OK(hipGetLastError());

    // dgrad (input gradient) — needed for every layer incl. the first
    // (the Function returns dx; unused grads are dropped by autograd)
    dx = torch::empty({batch, in_d}, opts);
    hipLaunchKernelGGL(mlp_dgrad_f32, dim3(n_blocks), dim3(256), 0, stream,
                       dy.data_ptr<float>(), y.data_ptr<float>(),
                       weights[l].data_ptr<float>(), dx.data_ptr<float>(), batch,
                       out_d, in_d, (int)acts[l]);
    HIP_OK(hipGetLastError());
    dy = dx;
  }

  std::vector<torch::Tensor> out;
  out.push_back(dx);
  for (int l = 0; l < L; ++l) out.push_back(dws[l]);
  for (int l = 0; l < L; ++l) out.push_back(dbs[l]);
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
                 double beta2, double eps, double weight_decay) {
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
    hipLaunchKernelGGL(fused_adam_kernel, dim3(a.n_tensors), dim3(256), 0,
                       current_stream(), a);
    HIP_OK(hipGetLastError());
  }
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
    hipLaunchKernelGGL(fused_polyak_kernel, dim3(a.n_tensors), dim3(256), 0,
                       current_stream(), a);
    HIP_OK(hipGetLastError());
  }
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("mlp_forward", &mlp_forward, "fused MLP forward (gfx950)");
  m.def("mlp_backward", &mlp_backward, "fused MLP backward (gfx950)");
  m.def("segmented_gae", &segmented_gae, "segmented GAE+returns scan (gfx950)");
  m.def("normalize", &normalize, "fused mean/std normalize (gfx950)");
  m.def("q_target", &q_target, "fused Q-learning target (gfx950)");
  m.def("fused_adam_", &fused_adam_, "fused multi-tensor Adam (gfx950)");
  m.def("fused_polyak_", &fused_polyak_, "fused multi-tensor Polyak (gfx950)");
}
