// Fused MLP forward/backward kernels for gfx950 (CDNA4).
//
// Replaces the reference's eager nn.Sequential forward + autograd
// backward (reference: src/rl_replicas/networks/mlp.py:29-41) for the
// tiny-MLP / launch-latency-bound regime this library lives in
// (SURVEY.md §7 "Tiny-tensor regime"): the whole multi-layer forward is
// ONE kernel launch; each workgroup owns a ROWS-row tile of the batch,
// ping-pongs layer activations between two padded LDS regions, and runs
// every Linear layer as MFMA tiles (v_mfma_f32_16x16x4_f32: exact fp32,
// guide §3) with bias+activation fused into the epilogue.
//
// Weight access is the latency hazard at these sizes (a scattered
// per-lane W read per MFMA serializes on L2 latency when the grid is
// small), so weights are staged through LDS:
//   mode 0 (narrow nets): the ENTIRE net's weights are staged once per
//     block into a padded LDS image ([out][(in+1)] per layer -> odd row
//     stride, bank-conflict-free for both W and W^T reads);
//   mode 1 (wide nets): each wave stages per-(output-tile, K-chunk)
//     sub-tiles of W into its private LDS slice, synchronized with
//     wave-local lgkmcnt waits (no cross-wave barriers).
// The host picks the mode from the LDS budget (dynamic shared memory).
//
// Backward = ONE merged kernel per layer (dgrad + wgrad/bias partials
// sharing the staged dZ tile) + a TWO-STAGE deterministic partial
// reduction (split-K via workspace instead of atomics so every run is
// bitwise reproducible — SURVEY.md §4).  Workspace layout is
// [block][all-layer elems] so the first reduction stage is layer-blind.
#include "common.h"

// ---------------------------------------------------------------------------
// helpers
// ---------------------------------------------------------------------------
template <int LDSW>
DEV_INLINE void load_tile(const float* __restrict__ src, float* dst_lds,
                          int row0, int batch, int width, int tid, int rows) {
  for (int idx = tid; idx < rows * width; idx += 256) {
    int r = idx / width, c = idx % width;
    float v = 0.f;
    int row = row0 + r;
    if (row < batch) v = src[(long)row * width + c];
    dst_lds[r * LDSW + c] = v;
  }
}

template <int LDSW>
DEV_INLINE void store_tile(float* __restrict__ dst, const float* src_lds,
                           int row0, int batch, int width, int tid, int rows) {
  for (int idx = tid; idx < rows * width; idx += 256) {
    int r = idx / width, c = idx % width;
    int row = row0 + r;
    if (row < batch) dst[(long)row * width + c] = src_lds[r * LDSW + c];
  }
}

// stage W[rows0..rows0+nrows)[0..ncols) -> lds image with row stride
// (ncols+1) — whole-block cooperative, caller barriers
DEV_INLINE void stage_weights_block(const float* __restrict__ W, float* wlds,
                                    int nrows, int ncols, int tid) {
  for (int idx = tid; idx < nrows * ncols; idx += 256) {
    int r = idx / ncols, c = idx % ncols;
    wlds[r * (ncols + 1) + c] = W[(long)r * ncols + c];
  }
}

// wave-local LDS write->read ordering (all writers are this wave's lanes)
DEV_INLINE void wave_lds_fence() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

#define KCHUNK 64  // K-chunk depth for mode-1 weight staging

// ---------------------------------------------------------------------------
// fused forward
// ---------------------------------------------------------------------------
template <int ROWS, int MAXW, bool BF16 = false>
__global__ __launch_bounds__(256) void fused_mlp_fwd_f32_t(
    MLPArgs args, const float* __restrict__ x, int save_hidden, int wstage_mode) {
  constexpr int LDSW = MAXW + 4;
  constexpr int RT = ROWS / 16;        // row tiles per block
  constexpr int JT_STRIDE = 4 / RT;    // waves sharing one row tile
  extern __shared__ float smem[];
  float* const buf0 = smem;
  float* const buf1 = smem + ROWS * LDSW;
  float* wlds = smem + 2 * ROWS * LDSW;  // mode 0: whole net; mode 1: per-wave
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * ROWS;
  const int wr0 = (wave % RT) * 16;
  const int jt0 = (wave / RT) * 16;
  // mode-1 per-wave W slice: [16][KCHUNK+2]
  float* wv = wlds + wave * 16 * (KCHUNK + 2);

  load_tile<LDSW>(x, buf0, row0, args.batch, args.dims[0], tid, ROWS);
  if (wstage_mode == 0) {
    int off = 0;
    for (int l = 0; l < args.n_layers; ++l) {
      stage_weights_block(args.w[l], wlds + off, args.dims[l + 1], args.dims[l], tid);
      off += args.dims[l + 1] * (args.dims[l] + 1);
    }
  }
  __syncthreads();

  const int i = lane & 15;
  const int k = lane >> 4;
  int cur = 0;
  int woff = 0;
  for (int l = 0; l < args.n_layers; ++l) {
    const int in_d = args.dims[l];
    const int out_d = args.dims[l + 1];
    const float* W = args.w[l];
    const float* B = args.b[l];
    const int act = args.acts[l];
    const int nxt = cur ^ 1;
    const float* buf_in = (cur == 0) ? buf0 : buf1;
    float* buf_out = (cur == 0) ? buf1 : buf0;
    const int wrow = in_d + 1;  // mode-0 padded W row stride

    for (int jt = jt0; jt < out_d; jt += 16 * JT_STRIDE) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      const int j = jt + i;
      const bool jok = j < out_d;
      if (wstage_mode == 0) {
        const float* wl = wlds + woff;
        if constexpr (BF16) {
          // bf16 compute: v_mfma_f32_16x16x32_bf16, fp32 accumulate;
          // fragments built from the fp32 LDS images with RNE converts
          for (int k0 = 0; k0 < in_d; k0 += 32) {
            bf16x8 af, bf;
            #pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int kk = k0 + k * 8 + e;
              af[e] = f32_to_bf16((kk < in_d) ? buf_in[(wr0 + i) * LDSW + kk] : 0.f);
              bf[e] = f32_to_bf16((jok && kk < in_d) ? wl[j * wrow + kk] : 0.f);
            }
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
          }
        } else {
          for (int k0 = 0; k0 < in_d; k0 += 4) {
            const int kk = k0 + k;
            float a = (kk < in_d) ? buf_in[(wr0 + i) * LDSW + kk] : 0.f;
            float bv = (jok && kk < in_d) ? wl[j * wrow + kk] : 0.f;
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
          }
        }
      } else if (wstage_mode == 2) {
        // direct global W reads (wide nets at large grids: the L2
        // misses hide behind cross-wave parallelism)
        for (int k0 = 0; k0 < in_d; k0 += 4) {
          const int kk = k0 + k;
          float a = (kk < in_d) ? buf_in[(wr0 + i) * LDSW + kk] : 0.f;
          float bv = (jok && kk < in_d) ? W[(long)j * in_d + kk] : 0.f;
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
        }
      } else {
        // per-wave staged K-chunks of W[jt..jt+16)[c0..c0+KCHUNK)
        for (int c0 = 0; c0 < in_d; c0 += KCHUNK) {
          const int clen = min(KCHUNK, in_d - c0);
          for (int idx = lane; idx < 16 * clen; idx += 64) {
            int r = idx / clen, c = idx % clen;
            float v = (jt + r < out_d) ? W[(long)(jt + r) * in_d + c0 + c] : 0.f;
            wv[r * (KCHUNK + 2) + c] = v;
          }
          wave_lds_fence();
          for (int k0 = 0; k0 < clen; k0 += 4) {
            const int kk = k0 + k;
            float a = (kk < clen) ? buf_in[(wr0 + i) * LDSW + c0 + kk] : 0.f;
            float bv = (kk < clen) ? wv[i * (KCHUNK + 2) + kk] : 0.f;
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
          }
          wave_lds_fence();  // reads done before next chunk overwrites
        }
      }
      if (jok) {
        const float bias = B[j];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = wr0 + (lane >> 4) * 4 + r;
          buf_out[row * LDSW + j] = act_apply(act, acc[r] + bias);
        }
      }
    }
    __syncthreads();

    const bool is_last = (l == args.n_layers - 1);
    if (is_last || save_hidden) {
      store_tile<LDSW>(args.h[l], buf_out, row0, args.batch, out_d, tid, ROWS);
    }
    cur = nxt;
    woff += out_d * wrow;
    // next layer writes bufs[cur^1] (fully consumed) and reads bufs[cur]
    // (fully written before the barrier above) -> one barrier per layer
  }
}

// ---------------------------------------------------------------------------
// merged backward layer: ONE kernel per layer computing
//   dZ = dY * act'(y)           (staged once in LDS)
//   dX = dZ @ W                 (row tiles, written to global)
//   dW_p = dZ^T @ X, db_p       (per-row-block partials to workspace)
// workspace layout: ws[block][layer-elems at layer_off]
// ---------------------------------------------------------------------------
template <int ROWS, int MAXW>
__global__ __launch_bounds__(256) void mlp_bwd_layer_f32_t(
    const float* __restrict__ dy, const float* __restrict__ y,
    const float* __restrict__ xin, const float* __restrict__ W,
    float* __restrict__ dx, float* __restrict__ workspace, long ws_stride,
    int batch, int out_d, int in_d, int act, int wstage_mode) {
  constexpr int LDSW = MAXW + 4;
  constexpr int RT = ROWS / 16;
  constexpr int JT_STRIDE = 4 / RT;
  extern __shared__ float smem[];
  float* dz = smem;
  float* xt = smem + ROWS * LDSW;
  float* wlds = smem + 2 * ROWS * LDSW;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * ROWS;
  const int wr0 = (wave % RT) * 16;
  const int jt0 = (wave / RT) * 16;
  float* wv = wlds + wave * KCHUNK * 18;  // mode-1 slice: [KCHUNK][16+2]
  // element-major workspace (ws[elem][block]): the latency-bound reduce
  // reads each element's partials CONTIGUOUSLY; writers scatter instead
  // (stores don't stall)
  float* wsp = workspace + blockIdx.x;
  const long WSN = gridDim.x;

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
  load_tile<LDSW>(xin, xt, row0, batch, in_d, tid, ROWS);
  if (wstage_mode == 0) {
    stage_weights_block(W, wlds, out_d, in_d, tid);
  }
  __syncthreads();

  const int i = lane & 15;
  const int k = lane >> 4;
  const int wrow = in_d + 1;

  // ---- dgrad: dX[b][j] = sum_k dZ[b][k] W[k][j] ----
  for (int jt = jt0; jt < in_d; jt += 16 * JT_STRIDE) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const int j = jt + i;
    const bool jok = j < in_d;
    if (wstage_mode == 0) {
      for (int k0 = 0; k0 < out_d; k0 += 4) {
        const int kk = k0 + k;
        float a = (kk < out_d) ? dz[(wr0 + i) * LDSW + kk] : 0.f;
        float bv = (jok && kk < out_d) ? wlds[kk * wrow + j] : 0.f;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
      }
    } else if (wstage_mode == 2) {
      for (int k0 = 0; k0 < out_d; k0 += 4) {
        const int kk = k0 + k;
        float a = (kk < out_d) ? dz[(wr0 + i) * LDSW + kk] : 0.f;
        float bv = (jok && kk < out_d) ? W[(long)kk * in_d + j] : 0.f;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
      }
    } else {
      // per-wave staged chunks of W[c0..c0+KCHUNK)[jt..jt+16)
      for (int c0 = 0; c0 < out_d; c0 += KCHUNK) {
        const int clen = min(KCHUNK, out_d - c0);
        for (int idx = lane; idx < clen * 16; idx += 64) {
          int r = idx / 16, c = idx % 16;
          float v = (jt + c < in_d) ? W[(long)(c0 + r) * in_d + jt + c] : 0.f;
          wv[r * 18 + c] = v;
        }
        wave_lds_fence();
        for (int k0 = 0; k0 < clen; k0 += 4) {
          const int kk = k0 + k;
          float a = (kk < clen) ? dz[(wr0 + i) * LDSW + c0 + kk] : 0.f;
          float bv = (kk < clen) ? wv[kk * 18 + i] : 0.f;
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
        }
        wave_lds_fence();
      }
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
          if (orow < out_d) wsp[((long)orow * in_d + col) * WSN] = acc[r];
        }
      }
    }
  }

  __syncthreads();
  for (int c = tid; c < out_d; c += 256) {
    float s = 0.f;
    #pragma unroll 4
    for (int r = 0; r < ROWS; ++r) s += dz[r * LDSW + c];
    wsp[((long)out_d * in_d + c) * WSN] = s;
  }
}

// ---------------------------------------------------------------------------
// whole-net fused backward (narrow nets): ONE kernel walks the layer
// chain backwards per 32-row block — dZ ping-pongs between two LDS
// tiles (the next layer's activation-grad is fused into the dgrad
// epilogue, read from the already-staged X tile), weights come from the
// whole-net LDS image, and wgrad/bias partials stream to the split-K
// workspace.  Replaces L per-layer launches with one.
// ---------------------------------------------------------------------------
// When mse_returns != nullptr the last layer's dZ is seeded directly
// from the value-MSE gradient 2*(v - ret)/B (final activation must be
// identity, out_d == 1 — the value-function case, ppo.py:283-287) and
// the per-block loss partial sum((v-ret)^2)/B goes to loss_partials.
// DO_FWD: the forward pass runs INSIDE this kernel (value-loop fast
// path): x is staged once, every layer's activations land in LDS
// (hlds) and never touch HBM; the MSE seed and the backward read them
// from LDS.  fp32 only; requires mse_returns.
template <int ROWS, bool BF16 = false, bool DO_FWD = false>
__global__ __launch_bounds__(256) void mlp_bwd_fused_f32_t(
    MLPBwdArgs args, const float* __restrict__ x, const float* __restrict__ dy,
    float* __restrict__ dx_out, float* __restrict__ workspace,
    const float* __restrict__ mse_returns, float* __restrict__ loss_partials,
    GaussSeedArgs gargs = GaussSeedArgs{}) {
  constexpr int MAXW = 64;
  constexpr int LDSW = MAXW + 4;
  constexpr int RT = ROWS / 16;
  constexpr int JT_STRIDE = 4 / RT;
  extern __shared__ float smem[];
  float* const dza = smem;
  float* const dzb = smem + ROWS * LDSW;
  float* const xt = smem + 2 * ROWS * LDSW;
  float* const wlds = smem + 3 * ROWS * LDSW;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * ROWS;
  const int wr0 = (wave % RT) * 16;
  const int jt0 = (wave / RT) * 16;
  float* wsp = workspace + blockIdx.x;  // element-major: ws[elem][block]
  const long WSN = gridDim.x;
  const int L = args.n_layers;

  // whole-net padded W image + layer offsets
  int woffs[MLP_MAX_LAYERS];
  int wtotal = 0;
  {
    int off = 0;
    for (int l = 0; l < L; ++l) {
      woffs[l] = off;
      stage_weights_block(args.w[l], wlds + off, args.dims[l + 1], args.dims[l], tid);
      off += args.dims[l + 1] * (args.dims[l] + 1);
    }
    wtotal = off;
  }
  // DO_FWD: per-layer activations live here, [L][ROWS][LDSW]
  float* const hlds = wlds + wtotal;
  if constexpr (DO_FWD) {
    const int fi = lane & 15;
    load_tile<LDSW>(x, xt, row0, args.batch, args.dims[0], tid, ROWS);
    __syncthreads();  // weights + x staged
    const float* fin = xt;
    for (int l = 0; l < L; ++l) {
      const int in_d = args.dims[l];
      const int out_d = args.dims[l + 1];
      const int wrow = in_d + 1;
      const float* wl = wlds + woffs[l];
      float* hl = hlds + l * ROWS * LDSW;
      const int fk = lane >> 4;
      for (int jt = jt0; jt < out_d; jt += 16 * JT_STRIDE) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        const int j = jt + fi;
        const bool jok = j < out_d;
        for (int k0 = 0; k0 < in_d; k0 += 4) {
          const int kk = k0 + fk;
          float a = (kk < in_d) ? fin[(wr0 + fi) * LDSW + kk] : 0.f;
          float bv = (jok && kk < in_d) ? wl[j * wrow + kk] : 0.f;
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
        }
        if (jok) {
          const float bias = args.b[l][j];
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int rr = wr0 + (lane >> 4) * 4 + r;
            hl[rr * LDSW + j] = act_apply(args.acts[l], acc[r] + bias);
          }
        }
      }
      __syncthreads();
      fin = hl;
    }
  }
  // Gaussian-PPO seed (DO_FWD only): loss + pending-KL + dlog_std
  // partials and the dmean dZ seed, all from the LDS-resident forward
  if (DO_FWD && gargs.actions != nullptr) {
    const float* outl = hlds + (L - 1) * ROWS * LDSW;
    const int D = args.dims[L];
    __shared__ float cbuf[64];       // per-row dlogp coeff (ROWS <= 64)
    __shared__ float lbuf[64];       // per-row loss
    __shared__ float kbuf[64];       // per-row pending KL
    const float inv_b = 1.f / (float)args.batch;
    if (tid < ROWS) {
      const int row = row0 + tid;
      float lossr = 0.f, klr = 0.f, c = 0.f;
      if (row < args.batch) {
        float base = 0.5f * (float)D * LOG_2PI;
        float q = 0.f;
        for (int d = 0; d < D; ++d) {
          const float ls = gargs.log_std[d];
          base += ls;
          const float sg = __expf(ls);
          const float diff =
              gargs.actions[(long)row * D + d] - outl[tid * LDSW + d];
          const float z = diff / sg;
          q += z * z;
        }
        const float logp = -0.5f * q - base;
        klr = gargs.old_logp[row] - logp;
        float lr_;
        c = dlogp_coeff(1, logp, gargs.old_logp[row], gargs.adv[row],
                        gargs.clip, &lr_) * inv_b;
        lossr = lr_ * inv_b;
      }
      cbuf[tid] = c;
      lbuf[tid] = lossr;
      kbuf[tid] = klr;
    }
    __syncthreads();
    // dZ seed = dmean (identity head)
    for (int idx = tid; idx < ROWS * D; idx += 256) {
      const int r = idx / D, d = idx % D;
      const int row = row0 + r;
      float v = 0.f;
      if (row < args.batch) {
        const float inv_s2 = __expf(-2.f * gargs.log_std[d]);
        const float diff =
            gargs.actions[(long)row * D + d] - outl[r * LDSW + d];
        v = cbuf[r] * diff * inv_s2;
      }
      dza[r * LDSW + d] = v;
    }
    // dlog_std partials -> ws pseudo-layer (fixed row order: deterministic)
    if (tid < D) {
      const float inv_s2 = __expf(-2.f * gargs.log_std[tid]);
      float acc = 0.f;
      for (int r = 0; r < ROWS; ++r) {
        const int row = row0 + r;
        if (row < args.batch) {
          const float diff =
              gargs.actions[(long)row * D + tid] - outl[r * LDSW + tid];
          acc += cbuf[r] * (diff * diff * inv_s2 - 1.f);
        }
      }
      wsp[(long)(gargs.dls_off + tid) * WSN] = acc;
    }
    if (tid == 0) {
      float sl = 0.f, sk = 0.f;
      for (int r = 0; r < ROWS; ++r) {
        sl += lbuf[r];
        sk += kbuf[r];
      }
      loss_partials[blockIdx.x] = sl;
      gargs.kl_partials[blockIdx.x] = sk;
    }
  } else
  // seed dZ for the last layer: dY * act'(final out), or the fused
  // value-MSE gradient when mse_returns is given
  {
    const int od = args.dims[L];
    const int act = args.acts[L - 1];
    const float* yl = args.h[L - 1];
    const float* yl_lds = hlds + (L - 1) * ROWS * LDSW;
    const float inv_b = 2.f / (float)args.batch;
    float loss_acc = 0.f;
    for (int idx = tid; idx < ROWS * od; idx += 256) {
      int r = idx / od, c = idx % od;
      int row = row0 + r;
      float v = 0.f;
      if (row < args.batch) {
        float yv;
        if constexpr (DO_FWD) {
          yv = yl_lds[r * LDSW + c];
        } else {
          yv = yl[(long)row * od + c];
        }
        if (mse_returns) {
          const float diff = yv - mse_returns[row];
          v = diff * inv_b;
          loss_acc += diff * diff;
        } else {
          v = dy[(long)row * od + c] * act_grad_from_y(act, yv);
        }
      }
      dza[r * LDSW + c] = v;
    }
    if (mse_returns) {
      // deterministic block loss partial (fixed wave order)
      loss_acc = wave_reduce_sum(loss_acc);
      __shared__ float lred[4];
      if ((tid & 63) == 0) lred[tid >> 6] = loss_acc;
      __syncthreads();
      if (tid == 0) {
        loss_partials[blockIdx.x] =
            ((lred[0] + lred[1]) + (lred[2] + lred[3])) / (float)args.batch;
      }
    }
  }
  __syncthreads();

  const int i = lane & 15;
  const int k = lane >> 4;
  const float* dz_cur = dza;
  float* dz_nxt = dzb;

  for (int l = L - 1; l >= 0; --l) {
    const int in_d = args.dims[l];
    const int out_d = args.dims[l + 1];
    const int wrow = in_d + 1;
    const float* wl = wlds + woffs[l];

    // X_l (input activations of layer l; post-act of layer l-1):
    // DO_FWD reads them straight from the LDS-resident forward
    const float* xl;
    if constexpr (DO_FWD) {
      xl = (l == 0) ? xt : hlds + (l - 1) * ROWS * LDSW;
    } else {
      load_tile<LDSW>(l == 0 ? x : args.h[l - 1], xt, row0, args.batch, in_d, tid, ROWS);
      __syncthreads();
      xl = xt;
    }

    // ---- wgrad partials: dW[i][j] = sum_r dZ[r][i] X[r][j] ----
    const int n_it = (out_d + 15) / 16;
    float* lw = wsp + (long)args.layer_off[l] * WSN;
    for (int it = wave; it < n_it; it += 4) {
      const int ii = it * 16 + i;
      for (int jt = 0; jt < in_d; jt += 16) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        if constexpr (BF16) {
          for (int k0 = 0; k0 < ROWS; k0 += 32) {
            bf16x8 af, bf;
            #pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int kk = k0 + k * 8 + e;
              af[e] = f32_to_bf16((kk < ROWS && ii < out_d) ? dz_cur[kk * LDSW + ii] : 0.f);
              bf[e] = f32_to_bf16((kk < ROWS && jt + i < in_d) ? xl[kk * LDSW + jt + i] : 0.f);
            }
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
          }
        } else {
          for (int k0 = 0; k0 < ROWS; k0 += 4) {
            float a = (ii < out_d) ? dz_cur[(k0 + k) * LDSW + ii] : 0.f;
            float bv = (jt + i < in_d) ? xl[(k0 + k) * LDSW + jt + i] : 0.f;
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
          }
        }
        const int col = jt + i;
        if (col < in_d) {
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int orow = it * 16 + (lane >> 4) * 4 + r;
            if (orow < out_d) lw[((long)orow * in_d + col) * WSN] = acc[r];
          }
        }
      }
    }
    // bias partial
    for (int c = tid; c < out_d; c += 256) {
      float s = 0.f;
      #pragma unroll 4
      for (int r = 0; r < ROWS; ++r) s += dz_cur[r * LDSW + c];
      lw[((long)out_d * in_d + c) * WSN] = s;
    }

    // ---- dgrad: dX[b][j] = sum_k dZ[b][k] W[k][j]; for l>0 the next
    // activation-grad is fused from the staged X tile ----
    const int prev_act = l > 0 ? args.acts[l - 1] : ACT_IDENTITY;
    for (int jt = jt0; jt < in_d; jt += 16 * JT_STRIDE) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      const int j = jt + i;
      const bool jok = j < in_d;
      if constexpr (BF16) {
        for (int k0 = 0; k0 < out_d; k0 += 32) {
          bf16x8 af, bf;
          #pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int kk = k0 + k * 8 + e;
            af[e] = f32_to_bf16((kk < out_d) ? dz_cur[(wr0 + i) * LDSW + kk] : 0.f);
            bf[e] = f32_to_bf16((jok && kk < out_d) ? wl[kk * wrow + j] : 0.f);
          }
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
        }
      } else {
        for (int k0 = 0; k0 < out_d; k0 += 4) {
          const int kk = k0 + k;
          float a = (kk < out_d) ? dz_cur[(wr0 + i) * LDSW + kk] : 0.f;
          float bv = (jok && kk < out_d) ? wl[kk * wrow + j] : 0.f;
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
        }
      }
      if (jok) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = wr0 + (lane >> 4) * 4 + r;
          if (l > 0) {
            const float g = act_grad_from_y(prev_act, xl[row * LDSW + j]);
            dz_nxt[row * LDSW + j] = acc[r] * g;
          } else {
            const int grow = row0 + row;
            if (grow < args.batch) dx_out[(long)grow * in_d + j] = acc[r];
          }
        }
      }
    }
    __syncthreads();
    // swap dz ping-pong
    const float* t = dz_cur;
    dz_cur = dz_nxt;
    dz_nxt = (float*)t;
  }
}

// ---------------------------------------------------------------------------
// wide-net single-layer kernels: 2D grids (row blocks x 64-col groups).
// The fused multi-layer kernel keeps every output column of a row tile
// in one block, which leaves most of the chip idle for [*,256,256,*]
// nets at minibatch-sized inputs (DDPG/TD3's 100-row hot path); these
// per-layer kernels spread the column dimension across blocks instead.
// ---------------------------------------------------------------------------
template <int ROWS>
__global__ __launch_bounds__(256) void mlp_layer_fwd_wide_f32(
    const float* __restrict__ x, const float* __restrict__ W,
    const float* __restrict__ B, float* __restrict__ out, int batch, int in_d,
    int out_d, int act) {
  constexpr int LDSW = 256 + 4;
  extern __shared__ float smem[];  // x tile [ROWS][LDSW]
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = blockIdx.x * ROWS;
  const int jt = blockIdx.y * 64 + wave * 16;  // one 16-col tile per wave

  load_tile<LDSW>(x, smem, row0, batch, in_d, tid, ROWS);
  __syncthreads();
  if (jt >= out_d) return;

  const int i = lane & 15;
  const int k = lane >> 4;
  const int j = jt + i;
  const bool jok = j < out_d;
  const float bias = jok ? B[j] : 0.f;
  const float* wrow = W + (long)j * in_d;
  const int in4 = in_d & ~3;
  const int in8 = in_d & ~7;
  for (int rt = 0; rt < ROWS; rt += 16) {
    // two independent accumulator chains double the MFMA/load ILP
    // (the K chain is otherwise a serial dependent sequence)
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    f32x4 acc2 = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = 0; k0 < in8; k0 += 8) {
      float bv0 = 0.f, bv1 = 0.f;
      if (jok) {
        // every lane of a 16-lane group reads its K element of the
        // broadcast W chunk (4x fewer load instructions than scalar)
        bv0 = wrow[k0 + k];
        bv1 = wrow[k0 + 4 + k];
      }
      const float a0 = smem[(rt + i) * LDSW + k0 + k];
      const float a1 = smem[(rt + i) * LDSW + k0 + 4 + k];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, bv0, acc, 0, 0, 0);
      acc2 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, bv1, acc2, 0, 0, 0);
    }
    for (int k0 = in8; k0 < in4; k0 += 4) {
      float bv = jok ? wrow[k0 + k] : 0.f;
      const float a = smem[(rt + i) * LDSW + k0 + k];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
    }
    if (in4 < in_d) {  // ragged K tail
      const int kk = in4 + k;
      float a = (kk < in_d) ? smem[(rt + i) * LDSW + kk] : 0.f;
      float bv = (jok && kk < in_d) ? wrow[kk] : 0.f;
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
    }
    acc[0] += acc2[0]; acc[1] += acc2[1]; acc[2] += acc2[2]; acc[3] += acc2[3];
    if (jok) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = row0 + rt + (lane >> 4) * 4 + r;
        if (row < batch) out[(long)row * out_d + j] = act_apply(act, acc[r] + bias);
      }
    }
  }
}

DEV_INLINE void dgrad_wide_body(
    float* __restrict__ smem, const float* __restrict__ dy,
    const float* __restrict__ y, const float* __restrict__ W,
    float* __restrict__ dx, int batch, int out_d, int in_d, int act,
    int bx, int by, int ROWS) {
  constexpr int LDSW = 256 + 4;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = bx * ROWS;
  const int jt = by * 64 + wave * 16;  // over in_d

  for (int idx = tid; idx < ROWS * out_d; idx += 256) {
    int r = idx / out_d, c = idx % out_d;
    int row = row0 + r;
    float v = 0.f;
    if (row < batch) {
      long g = (long)row * out_d + c;
      v = dy[g] * act_grad_from_y(act, y[g]);
    }
    smem[r * LDSW + c] = v;
  }
  __syncthreads();
  if (jt >= in_d) return;

  const int i = lane & 15;
  const int k = lane >> 4;
  const int j = jt + i;
  const bool jok = j < in_d;
  const int od8 = out_d & ~7;
  for (int rt = 0; rt < ROWS; rt += 16) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    f32x4 acc2 = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = 0; k0 < od8; k0 += 8) {
      const float a0 = smem[(rt + i) * LDSW + k0 + k];
      const float a1 = smem[(rt + i) * LDSW + k0 + 4 + k];
      float bv0 = jok ? W[(long)(k0 + k) * in_d + j] : 0.f;
      float bv1 = jok ? W[(long)(k0 + 4 + k) * in_d + j] : 0.f;
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, bv0, acc, 0, 0, 0);
      acc2 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, bv1, acc2, 0, 0, 0);
    }
    for (int k0 = od8; k0 < out_d; k0 += 4) {
      const int kk = k0 + k;
      float a = (kk < out_d) ? smem[(rt + i) * LDSW + kk] : 0.f;
      float bv = (jok && kk < out_d) ? W[(long)kk * in_d + j] : 0.f;
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
    }
    acc[0] += acc2[0]; acc[1] += acc2[1]; acc[2] += acc2[2]; acc[3] += acc2[3];
    if (jok) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = row0 + rt + (lane >> 4) * 4 + r;
        if (row < batch) dx[(long)row * in_d + j] = acc[r];
      }
    }
  }
}

template <int ROWS>
__global__ __launch_bounds__(256) void mlp_dgrad_wide_f32(
    const float* __restrict__ dy, const float* __restrict__ y,
    const float* __restrict__ W, float* __restrict__ dx, int batch, int out_d,
    int in_d, int act) {
  extern __shared__ float smem[];
  dgrad_wide_body(smem, dy, y, W, dx, batch, out_d, in_d, act, blockIdx.x,
                  blockIdx.y, ROWS);
}

// wgrad + bias partials, columns of dW's out dimension split over
// blockIdx.y (disjoint writes into the same per-row-block partial row)
DEV_INLINE void wgrad_wide_body(
    float* __restrict__ smem, const float* __restrict__ dy,
    const float* __restrict__ y, const float* __restrict__ xin,
    float* __restrict__ workspace, long ws_stride, int batch, int out_d,
    int in_d, int act, int bx, int by, int ROWS) {
  constexpr int LDSW = 256 + 4;
  constexpr int DZW = 64 + 4;
  float* xt = smem;
  float* dz = smem + ROWS * LDSW;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int row0 = bx * ROWS;
  const int og0 = by * 64;  // out-col group
  float* wsp = workspace + bx;  // element-major: ws[elem][block]
  const long WSN = gridDim.x;

  load_tile<LDSW>(xin, xt, row0, batch, in_d, tid, ROWS);
  const int og_w = min(64, out_d - og0);
  for (int idx = tid; idx < ROWS * og_w; idx += 256) {
    int r = idx / og_w, c = idx % og_w;
    int row = row0 + r;
    float v = 0.f;
    if (row < batch) {
      long g = (long)row * out_d + og0 + c;
      v = dy[g] * act_grad_from_y(act, y[g]);
    }
    dz[r * DZW + c] = v;
  }
  __syncthreads();

  const int i = lane & 15;
  const int k = lane >> 4;
  const int ii = wave * 16 + i;       // dz slice col (out index og0+ii)
  const bool iok = og0 + ii < out_d;
  for (int jt = 0; jt < in_d; jt += 16) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = 0; k0 < ROWS; k0 += 4) {
      float a = iok ? dz[(k0 + k) * DZW + ii] : 0.f;
      float bv = (jt + i < in_d) ? xt[(k0 + k) * LDSW + jt + i] : 0.f;
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
    }
    const int col = jt + i;
    if (col < in_d) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int orow = og0 + wave * 16 + (lane >> 4) * 4 + r;
        if (orow < out_d) wsp[((long)orow * in_d + col) * WSN] = acc[r];
      }
    }
  }

  __syncthreads();
  for (int c = tid; c < og_w; c += 256) {
    float s = 0.f;
    #pragma unroll 4
    for (int r = 0; r < ROWS; ++r) s += dz[r * DZW + c];
    wsp[((long)out_d * in_d + og0 + c) * WSN] = s;
  }
}

template <int ROWS>
__global__ __launch_bounds__(256) void mlp_wgrad_wide_f32(
    const float* __restrict__ dy, const float* __restrict__ y,
    const float* __restrict__ xin, float* __restrict__ workspace,
    long ws_stride, int batch, int out_d, int in_d, int act) {
  extern __shared__ float smem[];
  wgrad_wide_body(smem, dy, y, xin, workspace, ws_stride, batch, out_d, in_d,
                  act, blockIdx.x, blockIdx.y, ROWS);
}

// dgrad and wgrad of one layer are independent given dZ: one launch with
// blockIdx.z selecting the family doubles the resident block pool (the
// per-family grids are 16ish blocks at minibatch scale — far below the
// 256-CU chip) instead of serializing two launches on the stream
template <int ROWS>
__global__ __launch_bounds__(256) void mlp_bwd_wide_both_f32(
    const float* __restrict__ dy, const float* __restrict__ y,
    const float* __restrict__ xin, const float* __restrict__ W,
    float* __restrict__ dx, float* __restrict__ workspace, long ws_stride,
    int batch, int out_d, int in_d, int act) {
  extern __shared__ float smem[];
  if (blockIdx.z == 0) {
    if ((int)blockIdx.y * 64 < in_d)
      dgrad_wide_body(smem, dy, y, W, dx, batch, out_d, in_d, act, blockIdx.x,
                      blockIdx.y, ROWS);
  } else {
    if ((int)blockIdx.y * 64 < out_d)
      wgrad_wide_body(smem, dy, y, xin, workspace, ws_stride, batch, out_d,
                      in_d, act, blockIdx.x, blockIdx.y, ROWS);
  }
}

// host-side dispatch over the (ROWS, MAXW) instantiations — called from
// bindings.hip so template symbols stay in this translation unit
void launch_mlp_bwd_fused(const MLPBwdArgs& args, const float* x,
                          const float* dy, float* dx, float* ws,
                          const float* mse_returns, float* loss_partials,
                          size_t lds_bytes, int n_blocks, int compute_bf16,
                          hipStream_t stream, int rows, int do_fwd,
                          GaussSeedArgs gargs) {
  if (do_fwd) {  // fp32 only (host gates)
    if (rows == 16)
      hipLaunchKernelGGL((mlp_bwd_fused_f32_t<16, false, true>), dim3(n_blocks),
                         dim3(256), lds_bytes, stream, args, x, dy, dx, ws,
                         mse_returns, loss_partials, gargs);
    else
      hipLaunchKernelGGL((mlp_bwd_fused_f32_t<32, false, true>), dim3(n_blocks),
                         dim3(256), lds_bytes, stream, args, x, dy, dx, ws,
                         mse_returns, loss_partials, gargs);
    return;
  }
  if (rows == 16) {
    if (compute_bf16)
      hipLaunchKernelGGL((mlp_bwd_fused_f32_t<16, true>), dim3(n_blocks),
                         dim3(256), lds_bytes, stream, args, x, dy, dx, ws,
                         mse_returns, loss_partials);
    else
      hipLaunchKernelGGL((mlp_bwd_fused_f32_t<16, false>), dim3(n_blocks),
                         dim3(256), lds_bytes, stream, args, x, dy, dx, ws,
                         mse_returns, loss_partials);
    return;
  }
  if (compute_bf16)
    hipLaunchKernelGGL((mlp_bwd_fused_f32_t<32, true>), dim3(n_blocks),
                       dim3(256), lds_bytes, stream, args, x, dy, dx, ws,
                       mse_returns, loss_partials);
  else
    hipLaunchKernelGGL((mlp_bwd_fused_f32_t<32, false>), dim3(n_blocks),
                       dim3(256), lds_bytes, stream, args, x, dy, dx, ws,
                       mse_returns, loss_partials);
}

void launch_mlp_layer_fwd_wide(const float* x, const float* W, const float* B,
                               float* out, int batch, int in_d, int out_d,
                               int act, hipStream_t stream) {
  constexpr int ROWS = 32;
  dim3 g((batch + ROWS - 1) / ROWS, (out_d + 63) / 64), b(256);
  size_t lds = (size_t)ROWS * (256 + 4) * 4;
  hipLaunchKernelGGL((mlp_layer_fwd_wide_f32<ROWS>), g, b, lds, stream, x, W, B,
                     out, batch, in_d, out_d, act);
}

void launch_mlp_bwd_wide(const float* dy, const float* y, const float* xin,
                         const float* W, float* dx, float* ws, long ws_stride,
                         int batch, int out_d, int in_d, int act,
                         hipStream_t stream) {
  constexpr int ROWS = 32;
  const int rb = (batch + ROWS - 1) / ROWS;
  const int it = (in_d + 63) / 64, ot = (out_d + 63) / 64;
  const int yt = it > ot ? it : ot;
  size_t lds = (size_t)ROWS * (256 + 4 + 64 + 4) * 4;  // max of both bodies
  hipLaunchKernelGGL((mlp_bwd_wide_both_f32<ROWS>), dim3(rb, yt, 2), dim3(256),
                     lds, stream, dy, y, xin, W, dx, ws, ws_stride, batch,
                     out_d, in_d, act);
}

// dgrad chain only (input gradient without weight-grad partials) — the
// actor step's path through the frozen critic
void launch_mlp_dgrad_wide(const float* dy, const float* y, const float* W,
                           float* dx, int batch, int out_d, int in_d, int act,
                           hipStream_t stream) {
  constexpr int ROWS = 32;
  const int rb = (batch + ROWS - 1) / ROWS;
  size_t lds1 = (size_t)ROWS * (256 + 4) * 4;
  hipLaunchKernelGGL((mlp_dgrad_wide_f32<ROWS>), dim3(rb, (in_d + 63) / 64),
                     dim3(256), lds1, stream, dy, y, W, dx, batch, out_d, in_d,
                     act);
}

void launch_mlp_fwd(const MLPArgs& args, const float* x, int save_hidden,
                    int rows, int maxw, int n_blocks, int wstage_mode,
                    size_t lds_bytes, int compute_bf16, hipStream_t stream) {
  dim3 g(n_blocks), b(256);
  if (compute_bf16 && maxw == 64 && wstage_mode == 0) {
    if (rows == 16)
      hipLaunchKernelGGL((fused_mlp_fwd_f32_t<16, 64, true>), g, b, lds_bytes, stream, args, x, save_hidden, wstage_mode);
    else if (rows == 32)
      hipLaunchKernelGGL((fused_mlp_fwd_f32_t<32, 64, true>), g, b, lds_bytes, stream, args, x, save_hidden, wstage_mode);
    else
      hipLaunchKernelGGL((fused_mlp_fwd_f32_t<64, 64, true>), g, b, lds_bytes, stream, args, x, save_hidden, wstage_mode);
    return;
  }
  if (rows == 16 && maxw == 64)
    hipLaunchKernelGGL((fused_mlp_fwd_f32_t<16, 64>), g, b, lds_bytes, stream, args, x, save_hidden, wstage_mode);
  else if (rows == 32 && maxw == 64)
    hipLaunchKernelGGL((fused_mlp_fwd_f32_t<32, 64>), g, b, lds_bytes, stream, args, x, save_hidden, wstage_mode);
  else if (rows == 64 && maxw == 64)
    hipLaunchKernelGGL((fused_mlp_fwd_f32_t<64, 64>), g, b, lds_bytes, stream, args, x, save_hidden, wstage_mode);
  else
    hipLaunchKernelGGL((fused_mlp_fwd_f32_t<32, 256>), g, b, lds_bytes, stream, args, x, save_hidden, wstage_mode);
}

void launch_mlp_bwd_layer(const float* dy, const float* y, const float* xin,
                          const float* W, float* dx, float* ws, long ws_stride,
                          int batch, int out_d, int in_d, int act, int rows,
                          int maxw, int n_blocks, int wstage_mode,
                          size_t lds_bytes, hipStream_t stream) {
  dim3 g(n_blocks), b(256);
  if (rows == 32 && maxw == 64)
    hipLaunchKernelGGL((mlp_bwd_layer_f32_t<32, 64>), g, b, lds_bytes, stream, dy, y, xin, W, dx, ws, ws_stride, batch, out_d, in_d, act, wstage_mode);
  else if (rows == 64 && maxw == 64)
    hipLaunchKernelGGL((mlp_bwd_layer_f32_t<64, 64>), g, b, lds_bytes, stream, dy, y, xin, W, dx, ws, ws_stride, batch, out_d, in_d, act, wstage_mode);
  else
    hipLaunchKernelGGL((mlp_bwd_layer_f32_t<32, 256>), g, b, lds_bytes, stream, dy, y, xin, W, dx, ws, ws_stride, batch, out_d, in_d, act, wstage_mode);
}

// ---------------------------------------------------------------------------
// deterministic split-K partial reduction
// one-pass variant: 256 threads = 64 elements x 4 partial-quarters,
// combined through LDS in fixed order — one launch even at 125 partials
// reduce + Adam in one pass: identical fixed-order partial summation to
// mlp_grad_reduce_onepass_f32, but the summed gradient feeds the Adam
// update (same math as fused_adam_kernel) directly — no dw/db tensors.
__global__ __launch_bounds__(256) void mlp_grad_reduce_adam_f32(ReduceAdamArgs a) {
  if (a.gate && *a.gate == 0.f) return;
  __shared__ float red[256];
  int grand = 0;
  int base[MLP_MAX_LAYERS];
  for (int l = 0; l < a.n_layers; ++l) {
    base[l] = grand;
    grand += a.total[l];
  }
  const float s0 = a.step[0] + 1.f + a.step_delta;
  const float bc1 = 1.f - __powf(a.beta1, s0);
  const float bc2 = 1.f - __powf(a.beta2, s0);
  const int e_local = threadIdx.x & 63;
  const int quarter = threadIdx.x >> 6;
  const int chunk = (a.n_blocks + 3) / 4;
  const int p0 = quarter * chunk;
  const int p1 = min(p0 + chunk, a.n_blocks);
  for (int g0 = blockIdx.x * 64; g0 < grand; g0 += gridDim.x * 64) {
    const int g = g0 + e_local;
    float s = 0.f;
    if (g < grand) {
      for (int p = p0; p < p1; ++p) s += a.ws[(long)g * a.n_blocks + p];
    }
    red[threadIdx.x] = s;
    __syncthreads();
    if (quarter == 0 && g < grand) {
      float grad = (red[e_local] + red[64 + e_local]) +
                   (red[128 + e_local] + red[192 + e_local]);
      int l = 0;
      while (l + 1 < a.n_layers && g >= base[l + 1]) ++l;
      int idx = g - base[l];
      float *p, *m, *v;
      if (idx < a.wsize[l]) {
        p = a.pw[l]; m = a.mw[l]; v = a.vw[l];
      } else {
        idx -= a.wsize[l];
        p = a.pb[l]; m = a.mb[l]; v = a.vb[l];
      }
      if (a.weight_decay != 0.f) grad += a.weight_decay * p[idx];
      const float mi = a.beta1 * m[idx] + (1.f - a.beta1) * grad;
      const float vi = a.beta2 * v[idx] + (1.f - a.beta2) * grad * grad;
      m[idx] = mi;
      v[idx] = vi;
      p[idx] -= a.lr * (mi / bc1) / (sqrtf(vi / bc2) + a.eps);
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(256) void mlp_grad_reduce_onepass_f32(ReduceAllArgs a) {
  __shared__ float red[256];
  int grand = 0;
  int base[MLP_MAX_LAYERS];
  for (int l = 0; l < a.n_layers; ++l) {
    base[l] = grand;
    grand += a.total[l];
  }
  const int e_local = threadIdx.x & 63;
  const int quarter = threadIdx.x >> 6;
  const int chunk = (a.n_blocks + 3) / 4;
  const int p0 = quarter * chunk;
  const int p1 = min(p0 + chunk, a.n_blocks);
  for (int g0 = blockIdx.x * 64; g0 < grand; g0 += gridDim.x * 64) {
    const int g = g0 + e_local;
    float s = 0.f;
    if (g < grand) {
      for (int p = p0; p < p1; ++p) s += a.ws[(long)g * a.n_blocks + p];
    }
    red[threadIdx.x] = s;
    __syncthreads();
    if (quarter == 0 && g < grand) {
      const float tot = (red[e_local] + red[64 + e_local]) +
                        (red[128 + e_local] + red[192 + e_local]);
      int l = 0;
      while (l + 1 < a.n_layers && g >= base[l + 1]) ++l;
      const int idx = g - base[l];
      if (idx < a.wsize[l]) a.dw[l][idx] = tot;
      else a.db[l][idx - a.wsize[l]] = tot;
    }
    __syncthreads();
  }
}
