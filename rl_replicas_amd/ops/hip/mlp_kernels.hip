// Fused MLP forward/backward kernels for gfx950 (CDNA4).
//
// Replaces the reference's eager nn.Sequential forward + autograd
// backward (reference: src/rl_replicas/networks/mlp.py:29-41) for the
// tiny-MLP / launch-latency-bound regime this library lives in
// (SURVEY.md §7 "Tiny-tensor regime"): the whole multi-layer forward is
// ONE kernel launch; each workgroup owns a 64-row tile of the batch,
// ping-pongs layer activations between two padded LDS buffers, and runs
// every Linear layer as MFMA tiles (v_mfma_f32_16x16x4_f32: exact fp32,
// guide §3) with bias+activation fused into the epilogue.  The optional
// bf16 compute path (bench dtype) feeds the same fp32 LDS activations
// into v_mfma_f32_16x16x32_bf16 with fp32 accumulation.
//
// Backward = 3 kernels per layer (still far fewer than eager autograd's
// per-op launches), all recomputing dZ = dY * act'(y) from the saved
// post-activations in LDS:
//   - mlp_dgrad:          dX = dZ @ W           (64-row tiles, MFMA)
//   - mlp_wgrad_partial:  per-row-block partial dW = dZ^T X, db = colsum dZ
//   - mlp_grad_reduce:    deterministic fixed-order sum over row blocks
// (split-K via workspace instead of atomics so every run is bitwise
// reproducible — the framework's determinism contract).
#include "common.h"


typedef short bf16x8 __attribute__((ext_vector_type(8)));

// ---------------------------------------------------------------------------
// helpers
// ---------------------------------------------------------------------------
DEV_INLINE void load_tile_f32(const float* __restrict__ src, float* dst_lds,
                              int row0, int batch, int width, int tid) {
  // [MLP_ROWS x width] global -> LDS (padded stride), zero-filling rows
  // past the batch end
  for (int idx = tid; idx < MLP_ROWS * width; idx += 256) {
    int r = idx / width, c = idx % width;
    float v = 0.f;
    int row = row0 + r;
    if (row < batch) v = src[(long)row * width + c];
    dst_lds[r * MLP_LDSW + c] = v;
  }
}

DEV_INLINE void store_tile_f32(float* __restrict__ dst, const float* src_lds,
                               int row0, int batch, int width, int tid) {
  for (int idx = tid; idx < MLP_ROWS * width; idx += 256) {
    int r = idx / width, c = idx % width;
    int row = row0 + r;
    if (row < batch) dst[(long)row * width + c] = src_lds[r * MLP_LDSW + c];
  }
}

// One 16-col output tile of rows [wr0, wr0+16) x cols [jt, jt+16):
// A from LDS (padded stride), B = W[out][in] row-major, fp32 MFMA.
DEV_INLINE f32x4 gemm_tile_f32(const float* lds_in, const float* __restrict__ W,
                               int in_d, int out_d, int wr0, int jt, int lane) {
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int i = lane & 15;        // A row within tile / B col within tile
  const int k = lane >> 4;        // K sub-index (0..3)
  const int j = jt + i;
  const bool jok = j < out_d;
  for (int k0 = 0; k0 < in_d; k0 += 4) {
    const int kk = k0 + k;
    float a = (kk < in_d) ? lds_in[(wr0 + i) * MLP_LDSW + kk] : 0.f;
    float bv = (jok && kk < in_d) ? W[(long)j * in_d + kk] : 0.f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
  }
  return acc;
}

// ---------------------------------------------------------------------------
// fused forward
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 1) void fused_mlp_fwd_f32(
    MLPArgs args, const float* __restrict__ x, int save_hidden) {
  __shared__ float buf[2][MLP_ROWS * MLP_LDSW];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * MLP_ROWS;
  const int wr0 = wave * 16;  // 4 waves x 16 rows = 64 rows

  load_tile_f32(x, buf[0], row0, args.batch, args.dims[0], tid);
  __syncthreads();

  int cur = 0;
  for (int l = 0; l < args.n_layers; ++l) {
    const int in_d = args.dims[l];
    const int out_d = args.dims[l + 1];
    const float* W = args.w[l];
    const float* B = args.b[l];
    const int act = args.acts[l];
    const int nxt = cur ^ 1;

    for (int jt = 0; jt < out_d; jt += 16) {
      f32x4 acc = gemm_tile_f32(buf[cur], W, in_d, out_d, wr0, jt, lane);
      const int j = jt + (lane & 15);
      if (j < out_d) {
        const float bias = B[j];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = wr0 + (lane >> 4) * 4 + r;
          buf[nxt][row * MLP_LDSW + j] = act_apply(act, acc[r] + bias);
        }
      }
    }
    __syncthreads();

    const bool is_last = (l == args.n_layers - 1);
    if (is_last || save_hidden) {
      store_tile_f32(args.h[l], buf[nxt], row0, args.batch, out_d, tid);
    }
    cur = nxt;
    // next layer writes buf[cur^1] (fully consumed) and reads buf[cur]
    // (fully written before the barrier above) -> one barrier per layer
  }
}

// ---------------------------------------------------------------------------
// backward: dX = (dY * act'(y)) @ W
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 1) void mlp_dgrad_f32(
    const float* __restrict__ dy, const float* __restrict__ y,
    const float* __restrict__ W, float* __restrict__ dx,
    int batch, int out_d, int in_d, int act) {
  __shared__ float dz[MLP_ROWS * MLP_LDSW];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * MLP_ROWS;
  const int wr0 = wave * 16;

  // dZ tile = dY * act'(y), built elementwise on load
  for (int idx = tid; idx < MLP_ROWS * out_d; idx += 256) {
    int r = idx / out_d, c = idx % out_d;
    int row = row0 + r;
    float v = 0.f;
    if (row < batch) {
      long g = (long)row * out_d + c;
      v = dy[g] * act_grad_from_y(act, y[g]);
    }
    dz[r * MLP_LDSW + c] = v;
  }
  __syncthreads();

  // dX[b][j] = sum_k dZ[b][k] * W[k][j]; W row-major [out_d][in_d]
  const int i = lane & 15;
  const int k = lane >> 4;
  for (int jt = 0; jt < in_d; jt += 16) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const int j = jt + i;
    const bool jok = j < in_d;
    for (int k0 = 0; k0 < out_d; k0 += 4) {
      const int kk = k0 + k;
      float a = (kk < out_d) ? dz[(wr0 + i) * MLP_LDSW + kk] : 0.f;
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
}

// ---------------------------------------------------------------------------
// backward: per-row-block partials  dW_p = dZ^T @ X,  db_p = colsum(dZ)
// workspace layout per block p: [out_d*in_d weight partial | out_d bias partial]
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 1) void mlp_wgrad_partial_f32(
    const float* __restrict__ dy, const float* __restrict__ y,
    const float* __restrict__ xin, float* __restrict__ workspace,
    int batch, int out_d, int in_d, int act) {
  __shared__ float dz[MLP_ROWS * MLP_LDSW];
  __shared__ float xt[MLP_ROWS * MLP_LDSW];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * MLP_ROWS;
  float* wsp = workspace + (long)blockIdx.x * (out_d * in_d + out_d);

  for (int idx = tid; idx < MLP_ROWS * out_d; idx += 256) {
    int r = idx / out_d, c = idx % out_d;
    int row = row0 + r;
    float v = 0.f;
    if (row < batch) {
      long g = (long)row * out_d + c;
      v = dy[g] * act_grad_from_y(act, y[g]);
    }
    dz[r * MLP_LDSW + c] = v;
  }
  load_tile_f32(xin, xt, row0, batch, in_d, tid);
  __syncthreads();

  // dW[i][j] = sum_r dZ[r][i] * X[r][j]; MFMA over K = 64 rows.
  // 4 waves split the out_d dimension tiles.
  const int i = lane & 15;
  const int k = lane >> 4;
  const int n_it = (out_d + 15) / 16;
  for (int it = wave; it < n_it; it += 4) {
    const int ii = it * 16 + i;  // out index for A / col index for C
    for (int jt = 0; jt < in_d; jt += 16) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      for (int k0 = 0; k0 < MLP_ROWS; k0 += 4) {
        // A[i][k] = dZ[k0+k][it*16+i]  (transposed read, padded stride)
        float a = (ii < out_d) ? dz[(k0 + k) * MLP_LDSW + ii] : 0.f;
        float bv = (jt + i < in_d) ? xt[(k0 + k) * MLP_LDSW + jt + i] : 0.f;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
      }
      // C[row=out idx][col=in idx]: row = it*16 + (lane>>4)*4 + r, col = jt + i
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

  // bias partial: db[c] = sum_r dZ[r][c]
  __syncthreads();
  for (int c = tid; c < out_d; c += 256) {
    float s = 0.f;
    #pragma unroll 4
    for (int r = 0; r < MLP_ROWS; ++r) s += dz[r * MLP_LDSW + c];
    wsp[(long)out_d * in_d + c] = s;
  }
}

// fixed-order reduction over the n_blocks partials (deterministic)
__global__ void mlp_grad_reduce_f32(const float* __restrict__ workspace,
                                    float* __restrict__ dw, float* __restrict__ db,
                                    int n_blocks, int out_d, int in_d) {
  const int total = out_d * in_d + out_d;
  const long stride = total;
  for (int idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int p = 0; p < n_blocks; ++p) s += workspace[p * stride + idx];
    if (idx < out_d * in_d) dw[idx] = s;
    else db[idx - out_d * in_d] = s;
  }
}

// ---------------------------------------------------------------------------
// merged backward layer: ONE kernel per layer computing
//   dZ = dY * act'(y)           (staged once in LDS)
//   dX = dZ @ W                 (row tiles, written to global)
//   dW_p = dZ^T @ X, db_p       (per-row-block partials to workspace)
// halves the per-layer staging + launch count of the split kernels.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 1) void mlp_bwd_layer_f32(
    const float* __restrict__ dy, const float* __restrict__ y,
    const float* __restrict__ xin, const float* __restrict__ W,
    float* __restrict__ dx, float* __restrict__ workspace, int batch,
    int out_d, int in_d, int act) {
  __shared__ float dz[MLP_ROWS * MLP_LDSW];
  __shared__ float xt[MLP_ROWS * MLP_LDSW];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * MLP_ROWS;
  const int wr0 = wave * 16;
  float* wsp = workspace + (long)blockIdx.x * (out_d * in_d + out_d);

  for (int idx = tid; idx < MLP_ROWS * out_d; idx += 256) {
    int r = idx / out_d, c = idx % out_d;
    int row = row0 + r;
    float v = 0.f;
    if (row < batch) {
      long g = (long)row * out_d + c;
      v = dy[g] * act_grad_from_y(act, y[g]);
    }
    dz[r * MLP_LDSW + c] = v;
  }
  load_tile_f32(xin, xt, row0, batch, in_d, tid);
  __syncthreads();

  const int i = lane & 15;
  const int k = lane >> 4;

  // ---- dgrad: dX[b][j] = sum_k dZ[b][k] W[k][j] ----
  for (int jt = 0; jt < in_d; jt += 16) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const int j = jt + i;
    const bool jok = j < in_d;
    for (int k0 = 0; k0 < out_d; k0 += 4) {
      const int kk = k0 + k;
      float a = (kk < out_d) ? dz[(wr0 + i) * MLP_LDSW + kk] : 0.f;
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

  // ---- wgrad partials: dW[i][j] = sum_r dZ[r][i] X[r][j] ----
  const int n_it = (out_d + 15) / 16;
  for (int it = wave; it < n_it; it += 4) {
    const int ii = it * 16 + i;
    for (int jt = 0; jt < in_d; jt += 16) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      for (int k0 = 0; k0 < MLP_ROWS; k0 += 4) {
        float a = (ii < out_d) ? dz[(k0 + k) * MLP_LDSW + ii] : 0.f;
        float bv = (jt + i < in_d) ? xt[(k0 + k) * MLP_LDSW + jt + i] : 0.f;
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
    for (int r = 0; r < MLP_ROWS; ++r) s += dz[r * MLP_LDSW + c];
    wsp[(long)out_d * in_d + c] = s;
  }
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
