#include "hip/hip_runtime.h"
// Row softmax / log-softmax and LayerNorm for gfx950 (bf16 data, fp32
// accumulation).  Reference parity: src/operator/nn/softmax.cu (warp-level
// row softmax) and nn/layer_norm.cu (LayerNormFusedForwardKernel) —
// CDNA4-native: one wave64 (or one 256-thr block for wide rows) per row,
// online max+sum single pass where the row fits in registers, bf16x8
// vector loads, __shfl_xor wave reductions.
#include "common.h"

namespace mxa {

DEVINL float wave_red_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}
DEVINL float wave_red_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// block-level reductions via LDS (256 threads = 4 waves)
DEVINL float block_red_sum(float v, float* scratch) {
  v = wave_red_sum(v);
  int w = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) scratch[w] = v;
  __syncthreads();
  float r = scratch[0] + scratch[1] + scratch[2] + scratch[3];
  __syncthreads();
  return r;
}
DEVINL float block_red_max(float v, float* scratch) {
  v = wave_red_max(v);
  int w = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) scratch[w] = v;
  __syncthreads();
  float r = fmaxf(fmaxf(scratch[0], scratch[1]),
                  fmaxf(scratch[2], scratch[3]));
  __syncthreads();
  return r;
}

// ---------------------------------------------------------------------------
// softmax / log_softmax forward: one 256-thr block per row (D % 8 == 0)
// ---------------------------------------------------------------------------

// vlen: optional per-row valid length (masked/causal attention); columns
// >= vlen[row] get probability 0 (log: -inf -> stored as large negative).
// wave-per-row (4 rows / 256-thr block): shuffle-only reductions, no
// barriers, full lane activity at the hot widths (D=512 attention rows,
// D<=1024 hidden rows) — a 256-thread block per row left 3/4 of the
// lanes idle below D=2048.
template <bool LOG>
__global__ __launch_bounds__(256) void softmax_fwd(
    const bf16* __restrict__ x, bf16* __restrict__ y, long rows, int D,
    const int* __restrict__ vlen, float alpha) {
  const int w = threadIdx.x >> 6, l = threadIdx.x & 63;
  for (long row = blockIdx.x * 4 + w; row < rows;
       row += 4L * gridDim.x) {
    const bf16* xr = x + row * D;
    bf16* yr = y + row * D;
    const int V = vlen ? vlen[row] : D;
    float mx = -3.0e38f;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (i + j < V) mx = fmaxf(mx, bf2f(v[j]) * alpha);
    }
    mx = wave_red_max(mx);
    float sum = 0;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (i + j < V) sum += __expf(bf2f(v[j]) * alpha - mx);
    }
    sum = wave_red_sum(sum);
    float inv = 1.f / sum, lse = __logf(sum);
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xr[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        if (i + j >= V) {
          o[j] = LOG ? f2bf(-3.0e38f) : (bf16)0.f;
        } else {
          float e = bf2f(v[j]) * alpha - mx;
          o[j] = f2bf(LOG ? e - lse : __expf(e) * inv);
        }
      }
      *(bf16x8*)&yr[i] = o;
    }
  }
}

// softmax bwd: dx = (dy - sum(dy*y)) * y ; log_softmax bwd: dx = dy - exp(y)*sum(dy)
template <bool LOG>
__global__ __launch_bounds__(256) void softmax_bwd(
    const bf16* __restrict__ y, const bf16* __restrict__ dy,
    bf16* __restrict__ dx, long rows, int D, float alpha) {
  const int w = threadIdx.x >> 6, l = threadIdx.x & 63;
  for (long row = blockIdx.x * 4 + w; row < rows;
       row += 4L * gridDim.x) {
    const bf16* yr = y + row * D;
    const bf16* gr = dy + row * D;
    bf16* or_ = dx + row * D;
    float dot = 0;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 vy = *(const bf16x8*)&yr[i];
      bf16x8 vg = *(const bf16x8*)&gr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += LOG ? bf2f(vg[j]) : bf2f(vg[j]) * bf2f(vy[j]);
    }
    dot = wave_red_sum(dot);
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 vy = *(const bf16x8*)&yr[i];
      bf16x8 vg = *(const bf16x8*)&gr[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float yv = bf2f(vy[j]), gv = bf2f(vg[j]);
        o[j] = f2bf((LOG ? gv - __expf(yv) * dot : (gv - dot) * yv) * alpha);
      }
      *(bf16x8*)&or_[i] = o;
    }
  }
}

// ---------------------------------------------------------------------------
// Fused attention-probs softmax + dropout (wave-per-row).  One kernel
// produces p (kept for softmax bwd), pd = dropout(p) (the ctx-GEMM /
// dV operand) and the mask — replaces the softmax + dropout_fwd pair
// and the bwd-side pd recompute (three full passes over [BH,Tq,Tk]).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void softmax_dropout_fwd(
    const bf16* __restrict__ x, bf16* __restrict__ p,
    bf16* __restrict__ pd, unsigned char* __restrict__ mask, long rows,
    int D, const int* __restrict__ vlen, float alpha,
    const u64* __restrict__ seed_ptr, u64 salt, float pdrop,
    float dscale) {
  const int w = threadIdx.x >> 6, l = threadIdx.x & 63;
  u64 seed = seed_finalize(*seed_ptr + salt);  // graph-safe (see misc.hip)
  u32 thresh = (u32)(pdrop * 4294967296.0f);
  for (long row = blockIdx.x * 4 + w; row < rows;
       row += 4L * gridDim.x) {
    const bf16* xr = x + row * D;
    const int V = vlen ? vlen[row] : D;
    float mx = -3.0e38f;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (i + j < V) mx = fmaxf(mx, bf2f(v[j]) * alpha);
    }
    mx = wave_red_max(mx);
    float sum = 0;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (i + j < V) sum += __expf(bf2f(v[j]) * alpha - mx);
    }
    sum = wave_red_sum(sum);
    float inv = 1.f / sum;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xr[i];
      bf16x8 op, od;
      unsigned char mb[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float pv = (i + j < V)
                       ? __expf(bf2f(v[j]) * alpha - mx) * inv
                       : 0.f;
        u32 r = hash_u32(seed, row * D + i + j);
        bool keep = r >= thresh;
        mb[j] = keep;
        op[j] = f2bf(pv);
        od[j] = keep ? f2bf(pv * dscale) : (bf16)0.f;
      }
      *(bf16x8*)&p[row * D + i] = op;
      *(bf16x8*)&pd[row * D + i] = od;
      *(u64*)&mask[row * D + i] = *(const u64*)mb;
    }
  }
}

// ds = alpha * p * (dp - sum(dp * p)),  dp = mask * dscale * dpd
__global__ __launch_bounds__(256) void softmax_dropout_bwd(
    const bf16* __restrict__ p, const bf16* __restrict__ dpd,
    const unsigned char* __restrict__ mask, bf16* __restrict__ ds,
    long rows, int D, float alpha, float dscale) {
  const int w = threadIdx.x >> 6, l = threadIdx.x & 63;
  for (long row = blockIdx.x * 4 + w; row < rows;
       row += 4L * gridDim.x) {
    const bf16* pr = p + row * D;
    const bf16* gr = dpd + row * D;
    const unsigned char* mr = mask + row * D;
    bf16* or_ = ds + row * D;
    float dot = 0;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 vp = *(const bf16x8*)&pr[i];
      bf16x8 vg = *(const bf16x8*)&gr[i];
      u64 m8 = *(const u64*)&mr[i];
      const unsigned char* mb = (const unsigned char*)&m8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float dp = mb[j] ? bf2f(vg[j]) * dscale : 0.f;
        dot += dp * bf2f(vp[j]);
      }
    }
    dot = wave_red_sum(dot);
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 vp = *(const bf16x8*)&pr[i];
      bf16x8 vg = *(const bf16x8*)&gr[i];
      u64 m8 = *(const u64*)&mr[i];
      const unsigned char* mb = (const unsigned char*)&m8;
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float dp = mb[j] ? bf2f(vg[j]) * dscale : 0.f;
        o[j] = f2bf((dp - dot) * bf2f(vp[j]) * alpha);
      }
      *(bf16x8*)&or_[i] = o;
    }
  }
}

// ---------------------------------------------------------------------------
// LayerNorm forward/backward: block per row, saves mean/rstd for bwd
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void layernorm_fwd(
    const bf16* __restrict__ x, bf16* __restrict__ y,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ save_mean, float* __restrict__ save_rstd, long rows,
    int D, float eps) {
  const int w = threadIdx.x >> 6, l = threadIdx.x & 63;
  for (long row = blockIdx.x * 4 + w; row < rows;
       row += 4L * gridDim.x) {
    const bf16* xr = x + row * D;
    bf16* yr = y + row * D;
    float s = 0, ss = 0;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v[j]);
        s += f;
        ss += f * f;
      }
    }
    s = wave_red_sum(s);
    ss = wave_red_sum(ss);
    float mean = s / D;
    float var = ss / D - mean * mean;
    float rstd = rsqrtf(var > 0 ? var + eps : eps);
    if (l == 0) {
      save_mean[row] = mean;
      save_rstd[row] = rstd;
    }
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xr[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = (bf2f(v[j]) - mean) * rstd;
        o[j] = f2bf(xh * gamma[i + j] + beta[i + j]);
      }
      *(bf16x8*)&yr[i] = o;
    }
  }
}

__global__ __launch_bounds__(256) void layernorm_bwd_dx(
    const bf16* __restrict__ x, const bf16* __restrict__ dy,
    bf16* __restrict__ dx, const float* __restrict__ gamma,
    const float* __restrict__ save_mean, const float* __restrict__ save_rstd,
    long rows, int D) {
  const int w = threadIdx.x >> 6, l = threadIdx.x & 63;
  for (long row = blockIdx.x * 4 + w; row < rows;
       row += 4L * gridDim.x) {
    const bf16* xr = x + row * D;
    const bf16* gr = dy + row * D;
    bf16* or_ = dx + row * D;
    float mean = save_mean[row], rstd = save_rstd[row];
    float c1 = 0, c2 = 0;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 vx = *(const bf16x8*)&xr[i];
      bf16x8 vg = *(const bf16x8*)&gr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float gg = bf2f(vg[j]) * gamma[i + j];
        float xh = (bf2f(vx[j]) - mean) * rstd;
        c1 += gg;
        c2 += gg * xh;
      }
    }
    c1 = wave_red_sum(c1) / D;
    c2 = wave_red_sum(c2) / D;
    for (int i = l * 8; i < D; i += 64 * 8) {
      bf16x8 vx = *(const bf16x8*)&xr[i];
      bf16x8 vg = *(const bf16x8*)&gr[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float gg = bf2f(vg[j]) * gamma[i + j];
        float xh = (bf2f(vx[j]) - mean) * rstd;
        o[j] = f2bf((gg - c1 - xh * c2) * rstd);
      }
      *(bf16x8*)&or_[i] = o;
    }
  }
}

// column reduce for dgamma/dbeta: dgamma = sum dy*xhat, dbeta = sum dy
__global__ __launch_bounds__(256) void layernorm_bwd_dparams(
    const bf16* __restrict__ x, const bf16* __restrict__ dy,
    const float* __restrict__ save_mean, const float* __restrict__ save_rstd,
    float* __restrict__ dgamma, float* __restrict__ dbeta, long rows,
    int D) {
  // blocks over column chunks of 64; 256 thr = 32 rows x 8 chunks of 8
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int chunk = t & 7, rlane = t >> 3;
  const int c0 = cblk + chunk * 8;
  const bool active = c0 + 8 <= D;
  float sg[8] = {0}, sb[8] = {0};
  for (long m = blockIdx.y * 32 + rlane; active && m < rows;
       m += 32L * gridDim.y) {
    float mean = save_mean[m], rstd = save_rstd[m];
    bf16x8 vx = *(const bf16x8*)&x[m * D + c0];
    bf16x8 vg = *(const bf16x8*)&dy[m * D + c0];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float g = bf2f(vg[i]);
      float xh = (bf2f(vx[i]) - mean) * rstd;
      sg[i] += g * xh;
      sb[i] += g;
    }
  }
  __shared__ float red[2][32][64];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    red[0][rlane][chunk * 8 + i] = sg[i];
    red[1][rlane][chunk * 8 + i] = sb[i];
  }
  __syncthreads();
  if (t < 64) {
    float fg = 0, fb = 0;
#pragma unroll
    for (int r = 0; r < 32; ++r) {
      fg += red[0][r][t];
      fb += red[1][r][t];
    }
    int c = cblk + t;
    if (c < D) {
      atomicAdd(&dgamma[c], fg);
      atomicAdd(&dbeta[c], fb);
    }
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static int rows_grid(long rows) {
  long blocks = (rows + 3) / 4;  // 4 rows (one wave each) per block
  return (int)(blocks < 2048 ? (blocks < 1 ? 1 : blocks) : 2048);
}

void softmax_fwd_bf16(const bf16* x, bf16* y, long rows, int D, bool log_,
                      float alpha,
                      hipStream_t s) {
  if (log_)
   hipLaunchKernelGGL(( softmax_fwd<true>), dim3(rows_grid(rows)), dim3(256), 0, s, x, y, rows, D,
                                                      nullptr, alpha);
  else
   hipLaunchKernelGGL(( softmax_fwd<false>), dim3(rows_grid(rows)), dim3(256), 0, s, x, y, rows, D,
                                                       nullptr, alpha);
}

void softmax_masked_fwd_bf16(const bf16* x, bf16* y, long rows, int D,
                             const int* vlen, float alpha, hipStream_t s) {
 hipLaunchKernelGGL(( softmax_fwd<false>), dim3(rows_grid(rows)), dim3(256), 0, s, x, y, rows, D, vlen,
                                                     alpha);
}

void softmax_bwd_bf16(const bf16* y, const bf16* dy, bf16* dx, long rows,
                      int D, bool log_, float alpha, hipStream_t s) {
  if (log_)
   hipLaunchKernelGGL(( softmax_bwd<true>), dim3(rows_grid(rows)), dim3(256), 0, s, y, dy, dx, rows, D,
                                                      alpha);
  else
   hipLaunchKernelGGL(( softmax_bwd<false>), dim3(rows_grid(rows)), dim3(256), 0, s, y, dy, dx, rows, D,
                                                       alpha);
}

void softmax_dropout_fwd_bf16(const bf16* x, bf16* p, bf16* pd,
                              unsigned char* mask, long rows, int D,
                              const int* vlen, float alpha,
                              const u64* seed_ptr, u64 salt, float pdrop,
                              hipStream_t s) {
 hipLaunchKernelGGL(( softmax_dropout_fwd), dim3(rows_grid(rows)), dim3(256), 0, s, 
      x, p, pd, mask, rows, D, vlen, alpha, seed_ptr, salt, pdrop,
      1.f / (1.f - pdrop));
}

void softmax_dropout_bwd_bf16(const bf16* p, const bf16* dpd,
                              const unsigned char* mask, bf16* ds,
                              long rows, int D, float alpha, float pdrop,
                              hipStream_t s) {
 hipLaunchKernelGGL(( softmax_dropout_bwd), dim3(rows_grid(rows)), dim3(256), 0, s, 
      p, dpd, mask, ds, rows, D, alpha, 1.f / (1.f - pdrop));
}

void layernorm_fwd_bf16(const bf16* x, bf16* y, const float* gamma,
                        const float* beta, float* save_mean,
                        float* save_rstd, long rows, int D, float eps,
                        hipStream_t s) {
 hipLaunchKernelGGL(( layernorm_fwd), dim3(rows_grid(rows)), dim3(256), 0, s, x, y, gamma, beta,
                                                save_mean, save_rstd, rows,
                                                D, eps);
}

void layernorm_bwd_bf16(const bf16* x, const bf16* dy, bf16* dx,
                        const float* gamma, const float* save_mean,
                        const float* save_rstd, float* dgamma, float* dbeta,
                        long rows, int D, hipStream_t s) {
 hipLaunchKernelGGL(( layernorm_bwd_dx), dim3(rows_grid(rows)), dim3(256), 0, s, 
      x, dy, dx, gamma, save_mean, save_rstd, rows, D);
  HIP_CHECK(hipMemsetAsync(dgamma, 0, D * sizeof(float), s));
  HIP_CHECK(hipMemsetAsync(dbeta, 0, D * sizeof(float), s));
  int cb = cdiv_h(D, 64);
  long gy = min((rows + 255) / 256, (long)(768 / cb + 1));
  dim3 g(cb, (unsigned)(gy < 1 ? 1 : gy));
 hipLaunchKernelGGL(( layernorm_bwd_dparams), dim3(g), dim3(256), 0, s, x, dy, save_mean, save_rstd,
                                          dgamma, dbeta, rows, D);
}

}  // namespace mxa
