// Fused MLP forward/backward kernels for gfx950 (CDNA4).
//
// Replaces the reference's eager nn.Sequential forward + autograd
// backward (reference: src/rl_replicas/networks/mlp.py:29-41) for the
// tiny-MLP / launch-latency-bound regime this library lives in
// (SURVEY.md §7 "Tiny-tensor regime"): the whole multi-layer forward is
// ONE kernel launch; each workgroup owns a ROWS-row tile of the batch,
// ping-pongs layer activations between two padded LDS buffers, and runs
// every Linear layer as MFMA tiles (v_mfma_f32_16x16x4_f32: exact fp32,
// guide §3) with bias+activation fused into the epilogue.
//
// Kernels are templated on <ROWS, MAXW>:
//   MAXW = widest supported layer (64 for the on-policy nets, 256 for
//          the off-policy nets) -> LDS footprint 2*ROWS*(MAXW+4)*4 B,
//          so narrow nets run at multi-block-per-CU occupancy;
//   ROWS = batch rows per workgroup (32 fills the 256-CU chip at the
//          reference's 4000-row batches: 125 blocks; 64 for small
//          batches).  4 waves split (row-tile x output-tile) work.
//
// Backward = ONE merged kernel per layer (dgrad + wgrad/bias partials
// sharing the staged dZ tile) + ONE deterministic all-layer partial
// reduction (split-K via workspace instead of atomics so every run is
// bitwise reproducible — the framework's determinism contract,
// SURVEY.md §4).
#include "common.h"

// ---------------------------------------------------------------------------
// helpers
// ---------------------------------------------------------------------------
template <int ROWS, int LDSW>
DEV_INLINE void load_tile(const float* __restrict__ src, float* dst_lds,
                          int row0, int batch, int width, int tid) {
  for (int idx = tid; idx < ROWS * width; idx += 256) {
    int r = idx / width, c = idx % width;
    float v = 0.f;
    int row = row0 + r;
    if (row < batch) v = src[(long)row * width + c];
    dst_lds[r * LDSW + c] = v;
  }
}

template <int ROWS, int LDSW>
DEV_INLINE void store_tile(float* __restrict__ dst, const float* src_lds,
                           int row0, int batch, int width, int tid) {
  for (int idx = tid; idx < ROWS * width; idx += 256) {
    int r = idx / width, c = idx % width;
    int row = row0 + r;
    if (row < batch) dst[(long)row * width + c] = src_lds[r * LDSW + c];
  }
}

// ---------------------------------------------------------------------------
// fused forward
// ---------------------------------------------------------------------------
template <int ROWS, int MAXW>
__global__ __launch_bounds__(256) void fused_mlp_fwd_f32_t(
    MLPArgs args, const float* __restrict__ x, int save_hidden) {
  constexpr int LDSW = MAXW + 4;
  constexpr int RT = ROWS / 16;        // row tiles per block
  constexpr int JT_STRIDE = 4 / RT;    // waves sharing one row tile
  __shared__ float buf[2][ROWS * LDSW];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * ROWS;
  const int wr0 = (wave % RT) * 16;
  const int jt0 = (wave / RT) * 16;

  load_tile<ROWS, LDSW>(x, buf[0], row0, args.batch, args.dims[0], tid);
  __syncthreads();

  const int i = lane & 15;
  const int k = lane >> 4;
  int cur = 0;
  for (int l = 0; l < args.n_layers; ++l) {
    const int in_d = args.dims[l];
    const int out_d = args.dims[l + 1];
    const float* W = args.w[l];
    const float* B = args.b[l];
    const int act = args.acts[l];
    const int nxt = cur ^ 1;

    for (int jt = jt0; jt < out_d; jt += 16 * JT_STRIDE) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      const int j = jt + i;
      const bool jok = j < out_d;
      for (int k0 = 0; k0 < in_d; k0 += 4) {
        const int kk = k0 + k;
        float a = (kk < in_d) ? buf[cur][(wr0 + i) * LDSW + kk] : 0.f;
        float bv = (jok && kk < in_d) ? W[(long)j * in_d + kk] : 0.f;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
      }
      if (jok) {
        const float bias = B[j];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = wr0 + (lane >> 4) * 4 + r;
          buf[nxt][row * LDSW + j] = act_apply(act, acc[r] + bias);
        }
      }
    }
    __syncthreads();

    const bool is_last = (l == args.n_layers - 1);
    if (is_last || save_hidden) {
      store_tile<ROWS, LDSW>(args.h[l], buf[nxt], row0, args.batch, out_d, tid);
    }
    cur = nxt;
    // next layer writes buf[cur^1] (fully consumed) and reads buf[cur]
    // (fully written before the barrier above) -> one barrier per layer
  }
}

// ---------------------------------------------------------------------------
// merged backward layer: ONE kernel per layer computing
//   dZ = dY * act'(y)           (staged once in LDS)
//   dX = dZ @ W                 (row tiles, written to global)
//   dW_p = dZ^T @ X, db_p       (per-row-block partials to workspace)
// ---------------------------------------------------------------------------
template <int ROWS, int MAXW>
__global__ __launch_bounds__(256) void mlp_bwd_layer_f32_t(
    const float* __restrict__ dy, const float* __restrict__ y,
    const float* __restrict__ xin, const float* __restrict__ W,
    float* __restrict__ dx, float* __restrict__ workspace, int batch,
    int out_d, int in_d, int act) {
  constexpr int LDSW = MAXW + 4;
  constexpr int RT = ROWS / 16;
  constexpr int JT_STRIDE = 4 / RT;
  __shared__ float dz[ROWS * LDSW];
  __shared__ float xt[ROWS * LDSW];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * ROWS;
  const int wr0 = (wave % RT) * 16;
  const int jt0 = (wave / RT) * 16;
  float* wsp = workspace + (long)blockIdx.x * (out_d * in_d + out_d);

  for (int idx = tid; idx < ROWS * out_d; idx += 256) {
    int r = idx / out_d, c = idx % out_d;
    int row = row0 + r;
    float v = 0.f;
    if (row < batch) {
      long g = (long)row * out_d + c;
      v = dy[g] * act_grad_from_y(act, y[g]);
    }
    dz[r * LDSW + c] = v;
  }
  load_tile<ROWS, LDSW>(xin, xt, row0, batch, in_d, tid);
  __syncthreads();

  const int i = lane & 15;
  const int k = lane >> 4;

  // ---- dgrad: dX[b][j] = sum_k dZ[b][k] W[k][j] ----
  for (int jt = jt0; jt < in_d; jt += 16 * JT_STRIDE) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const int j = jt + i;
    const bool jok = j < in_d;
    for (int k0 = 0; k0 < out_d; k0 += 4) {
      const int kk = k0 + k;
      float a = (kk < out_d) ? dz[(wr0 + i) * LDSW + kk] : 0.f;
      float bv = (jok && kk < out_d) ? W[(long)kk * in_d + j] : 0.f;
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
    }
    if (jok) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = row0 + wr0 + (lane >> 4) * 4 + r;
        if (row < batch) dx[(long)row * in_d + jt + i] = acc[r];
      }
    }
  }

  // ---- wgrad partials: dW[i][j] = sum_r dZ[r][i] X[r][j], K = ROWS ----
  const int n_it = (out_d + 15) / 16;
  for (int it = wave; it < n_it; it += 4) {
    const int ii = it * 16 + i;
    for (int jt = 0; jt < in_d; jt += 16) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      for (int k0 = 0; k0 < ROWS; k0 += 4) {
        float a = (ii < out_d) ? dz[(k0 + k) * LDSW + ii] : 0.f;
        float bv = (jt + i < in_d) ? xt[(k0 + k) * LDSW + jt + i] : 0.f;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
      }
      const int col = jt + i;
      if (col < in_d) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int orow = it * 16 + (lane >> 4) * 4 + r;
          if (orow < out_d) wsp[(long)orow * in_d + col] = acc[r];
        }
      }
    }
  }

  __syncthreads();
  for (int c = tid; c < out_d; c += 256) {
    float s = 0.f;
    #pragma unroll 4
    for (int r = 0; r < ROWS; ++r) s += dz[r * LDSW + c];
    wsp[(long)out_d * in_d + c] = s;
  }
}

// host-side dispatch over the (ROWS, MAXW) instantiations — called from
// bindings.hip so template symbols stay in this translation unit
void launch_mlp_fwd(const MLPArgs& args, const float* x, int save_hidden,
                    int rows, int maxw, int n_blocks, hipStream_t stream) {
  dim3 g(n_blocks), b(256);
  if (rows == 32 && maxw == 64)
    hipLaunchKernelGGL((fused_mlp_fwd_f32_t<32, 64>), g, b, 0, stream, args, x, save_hidden);
  else if (rows == 64 && maxw == 64)
    hipLaunchKernelGGL((fused_mlp_fwd_f32_t<64, 64>), g, b, 0, stream, args, x, save_hidden);
  else if (rows == 32 && maxw == 256)
    hipLaunchKernelGGL((fused_mlp_fwd_f32_t<32, 256>), g, b, 0, stream, args, x, save_hidden);
  else
    hipLaunchKernelGGL((fused_mlp_fwd_f32_t<64, 256>), g, b, 0, stream, args, x, save_hidden);
}

void launch_mlp_bwd_layer(const float* dy, const float* y, const float* xin,
                          const float* W, float* dx, float* ws, int batch,
                          int out_d, int in_d, int act, int rows, int maxw,
                          int n_blocks, hipStream_t stream) {
  dim3 g(n_blocks), b(256);
  if (rows == 32 && maxw == 64)
    hipLaunchKernelGGL((mlp_bwd_layer_f32_t<32, 64>), g, b, 0, stream, dy, y, xin, W, dx, ws, batch, out_d, in_d, act);
  else if (rows == 64 && maxw == 64)
    hipLaunchKernelGGL((mlp_bwd_layer_f32_t<64, 64>), g, b, 0, stream, dy, y, xin, W, dx, ws, batch, out_d, in_d, act);
  else if (rows == 32 && maxw == 256)
    hipLaunchKernelGGL((mlp_bwd_layer_f32_t<32, 256>), g, b, 0, stream, dy, y, xin, W, dx, ws, batch, out_d, in_d, act);
  else
    hipLaunchKernelGGL((mlp_bwd_layer_f32_t<64, 256>), g, b, 0, stream, dy, y, xin, W, dx, ws, batch, out_d, in_d, act);
}

// all-layer deterministic partial reduction: one launch per backward.
// workspace holds per-layer segments of n_blocks partials each;
// fixed (s0+s1)+(s2+s3) accumulation order -> bitwise reproducible.
__global__ void mlp_grad_reduce_all_f32(ReduceAllArgs a) {
  // flatten all layers' elements into one grid-stride loop
  int grand = 0;
  int base[MLP_MAX_LAYERS];
  for (int l = 0; l < a.n_layers; ++l) {
    base[l] = grand;
    grand += a.total[l];
  }
  for (int g = blockIdx.x * blockDim.x + threadIdx.x; g < grand;
       g += gridDim.x * blockDim.x) {
    int l = 0;
    while (l + 1 < a.n_layers && g >= base[l + 1]) ++l;
    const int idx = g - base[l];
    const long stride = a.total[l];
    const float* ws = a.ws[l];
    float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
    int p = 0;
    for (; p + 3 < a.n_blocks; p += 4) {
      s0 += ws[(p + 0) * stride + idx];
      s1 += ws[(p + 1) * stride + idx];
      s2 += ws[(p + 2) * stride + idx];
      s3 += ws[(p + 3) * stride + idx];
    }
    for (; p < a.n_blocks; ++p) s0 += ws[p * stride + idx];
    const float s = (s0 + s1) + (s2 + s3);
    if (idx < a.wsize[l]) a.dw[l][idx] = s;
    else a.db[l][idx - a.wsize[l]] = s;
  }
}
