// Fused BatchNorm (NHWC, bf16 data / fp32 stats) and pooling kernels for
// gfx950.  Reference parity: src/operator/nn/batch_norm.cu (+cudnn_batch_norm)
// and nn/pooling.cu — re-designed CDNA4-native: NHWC channel-contiguous
// bf16x8 vector loads (guide G13), fp32 accumulation, fused ReLU epilogue,
// per-channel partial sums reduced with fp32 atomics (memory-bound ops; the
// 8 TB/s HBM3E is the roofline, so one pass per tensor read is the goal).
#include "common.h"

namespace mxa {

// ---------------------------------------------------------------------------
// BatchNorm forward: stats (partial) -> finalize -> apply(+relu)
// x is [M][C] with M = N*H*W.  Blocks: 256 thr = 32 rows x 8 col-chunks of
// 8 channels; grid.x covers channel blocks of 64, grid.y strides rows.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void bn_stats_partial(
    const bf16* __restrict__ x, float* __restrict__ psum,
    float* __restrict__ psumsq, long M, int C) {
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int chunk = t & 7;           // 8 chunks x 8 channels = 64 channels
  const int rlane = t >> 3;          // 32 row lanes
  const int c0 = cblk + chunk * 8;
  const bool active = c0 + 8 <= C;
  float s[8] = {0}, ss[8] = {0};
  for (long m = blockIdx.y * 32 + rlane; active && m < M;
       m += 32L * gridDim.y) {
    bf16x8 v = *(const bf16x8*)&x[m * C + c0];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float f = bf2f(v[i]);
      s[i] += f;
      ss[i] += f * f;
    }
  }
  // reduce across the 32 row-lanes holding the same channels via LDS
  __shared__ float red[2][32][64];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    red[0][rlane][chunk * 8 + i] = s[i];
    red[1][rlane][chunk * 8 + i] = ss[i];
  }
  __syncthreads();
  // 64 threads accumulate 64 channels (one atomic per block per channel)
  if (t < 64) {
    float fs = 0, fss = 0;
#pragma unroll
    for (int r = 0; r < 32; ++r) {
      fs += red[0][r][t];
      fss += red[1][r][t];
    }
    int c = cblk + t;
    if (c < C) {
      atomicAdd(&psum[c], fs);
      atomicAdd(&psumsq[c], fss);
    }
  }
}

__global__ void bn_finalize(const float* __restrict__ psum,
                            const float* __restrict__ psumsq,
                            float* __restrict__ save_mean,
                            float* __restrict__ save_rstd,
                            float* __restrict__ running_mean,
                            float* __restrict__ running_var, long M, int C,
                            int gy, float eps, float momentum,
                            int update_running, int nslots,
                            long slot_stride) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, q = 0.f;
  for (int k = 0; k < nslots; ++k) {
    s += psum[(long)k * slot_stride + c];
    q += psumsq[(long)k * slot_stride + c];
  }
  float mean = s / M;
  float var = q / M - mean * mean;
  if (var < 0) var = 0;
  save_mean[c] = mean;
  save_rstd[c] = rsqrtf(var + eps);
  if (update_running) {
    running_mean[c] = momentum * running_mean[c] + (1.f - momentum) * mean;
    float ub = M > 1 ? var * M / (M - 1) : var;  // unbiased (torch parity)
    running_var[c] = momentum * running_var[c] + (1.f - momentum) * ub;
  }
}

// Apply-family geometry: column stripes (fixed 8 channels per thread,
// rows strided) so every per-channel coefficient is computed ONCE and
// hoisted out of the row loop — the flat grid-stride form reloaded 6
// scalars per 8 elements and ran at 1/3 of the HBM roofline.
template <bool RELU>
__global__ __launch_bounds__(256) void bn_apply(
    const bf16* __restrict__ x, bf16* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    long M, int C) {
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int c0 = cblk + (t & 7) * 8;
  if (c0 + 8 > C) return;
  float a[8], b[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int c = c0 + i;
    a[i] = rstd[c] * gamma[c];
    b[i] = beta[c] - mean[c] * a[i];
  }
  for (long m = blockIdx.y * 32 + (t >> 3); m < M; m += 32L * gridDim.y) {
    bf16x8 v = *(const bf16x8*)&x[m * C + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]) * a[j] + b[j];
      if (RELU) f = f > 0 ? f : 0;
      o[j] = f2bf(f);
    }
    *(bf16x8*)&y[m * C + c0] = o;
  }
}

// junction form: y = relu(bn(x) + residual) — fuses the residual add
// and the block-output ReLU into the BN epilogue (one HBM round trip
// instead of three; backward reconstructs the mask from y > 0).
__global__ __launch_bounds__(256) void bn_apply_add(
    const bf16* __restrict__ x, const bf16* __restrict__ r,
    bf16* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, long M, int C) {
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int c0 = cblk + (t & 7) * 8;
  if (c0 + 8 > C) return;
  float a[8], b[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int c = c0 + i;
    a[i] = rstd[c] * gamma[c];
    b[i] = beta[c] - mean[c] * a[i];
  }
  for (long m = blockIdx.y * 32 + (t >> 3); m < M; m += 32L * gridDim.y) {
    bf16x8 v = *(const bf16x8*)&x[m * C + c0];
    bf16x8 rv = *(const bf16x8*)&r[m * C + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]) * a[j] + b[j] + bf2f(rv[j]);
      o[j] = f2bf(f > 0 ? f : 0);
    }
    *(bf16x8*)&y[m * C + c0] = o;
  }
}

__global__ __launch_bounds__(256) void bn_infer_add(
    const bf16* __restrict__ x, const bf16* __restrict__ r,
    bf16* __restrict__ y, const float* __restrict__ rmean,
    const float* __restrict__ rvar, const float* __restrict__ gamma,
    const float* __restrict__ beta, long M, int C, float eps) {
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int c0 = cblk + (t & 7) * 8;
  if (c0 + 8 > C) return;
  float a[8], b[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int c = c0 + i;
    a[i] = rsqrtf(rvar[c] + eps) * gamma[c];
    b[i] = beta[c] - rmean[c] * a[i];
  }
  for (long m = blockIdx.y * 32 + (t >> 3); m < M; m += 32L * gridDim.y) {
    bf16x8 v = *(const bf16x8*)&x[m * C + c0];
    bf16x8 rv = *(const bf16x8*)&r[m * C + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]) * a[j] + b[j] + bf2f(rv[j]);
      o[j] = f2bf(f > 0 ? f : 0);
    }
    *(bf16x8*)&y[m * C + c0] = o;
  }
}

// inference: normalize with running stats directly
template <bool RELU>
__global__ __launch_bounds__(256) void bn_infer(
    const bf16* __restrict__ x, bf16* __restrict__ y,
    const float* __restrict__ rmean, const float* __restrict__ rvar,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    long M, int C, float eps) {
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int c0 = cblk + (t & 7) * 8;
  if (c0 + 8 > C) return;
  float a[8], b[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int c = c0 + i;
    a[i] = rsqrtf(rvar[c] + eps) * gamma[c];
    b[i] = beta[c] - rmean[c] * a[i];
  }
  for (long m = blockIdx.y * 32 + (t >> 3); m < M; m += 32L * gridDim.y) {
    bf16x8 v = *(const bf16x8*)&x[m * C + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]) * a[j] + b[j];
      if (RELU) f = f > 0 ? f : 0;
      o[j] = f2bf(f);
    }
    *(bf16x8*)&y[m * C + c0] = o;
  }
}

// backward reduction: dbeta = sum dy_eff, dgamma = sum dy_eff * xhat
// (RELU fused: dy_eff = dy * (y > 0), y reconstructed from xhat)
template <bool RELU>
__global__ __launch_bounds__(256) void bn_bwd_reduce(
    const bf16* __restrict__ x, const bf16* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ dbeta, float* __restrict__ dgamma, long M, int C) {
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int chunk = t & 7, rlane = t >> 3;
  const int c0 = cblk + chunk * 8;
  const bool active = c0 + 8 <= C;
  float sb[8] = {0}, sg[8] = {0};
  for (long m = blockIdx.y * 32 + rlane; active && m < M;
       m += 32L * gridDim.y) {
    bf16x8 vx = *(const bf16x8*)&x[m * C + c0];
    bf16x8 vdy = *(const bf16x8*)&dy[m * C + c0];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      int c = c0 + i;
      float xh = (bf2f(vx[i]) - mean[c]) * rstd[c];
      float g = bf2f(vdy[i]);
      if (RELU && (xh * gamma[c] + beta[c]) <= 0) g = 0;
      sb[i] += g;
      sg[i] += g * xh;
    }
  }
  __shared__ float red[2][32][64];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    red[0][rlane][chunk * 8 + i] = sb[i];
    red[1][rlane][chunk * 8 + i] = sg[i];
  }
  __syncthreads();
  if (t < 64) {
    float fb = 0, fg = 0;
#pragma unroll
    for (int r = 0; r < 32; ++r) {
      fb += red[0][r][t];
      fg += red[1][r][t];
    }
    int c = cblk + t;
    if (c < C) {
      atomicAdd(&dbeta[c], fb);
      atomicAdd(&dgamma[c], fg);
    }
  }
}

template <bool RELU>
__global__ __launch_bounds__(256) void bn_bwd_apply(
    const bf16* __restrict__ x, const bf16* __restrict__ dy,
    bf16* __restrict__ dx, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, const float* __restrict__ dbeta,
    const float* __restrict__ dgamma, long M, int C) {
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int c0 = cblk + (t & 7) * 8;
  if (c0 + 8 > C) return;
  float invM = 1.f / (float)M;
  // dx = g*A + x*B + K (g relu-masked via y = x*a + b when fused)
  float A[8], B[8], K[8], a[8], b[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int c = c0 + i;
    A[i] = gamma[c] * rstd[c];
    float e = rstd[c] * dgamma[c] * invM * A[i];
    B[i] = -e;
    K[i] = mean[c] * e - dbeta[c] * invM * A[i];
    a[i] = rstd[c] * gamma[c];
    b[i] = beta[c] - mean[c] * a[i];
  }
  for (long m = blockIdx.y * 32 + (t >> 3); m < M; m += 32L * gridDim.y) {
    bf16x8 vx = *(const bf16x8*)&x[m * C + c0];
    bf16x8 vdy = *(const bf16x8*)&dy[m * C + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xv = bf2f(vx[j]);
      float g = bf2f(vdy[j]);
      if (RELU && (xv * a[j] + b[j]) <= 0) g = 0;
      o[j] = f2bf(g * A[j] + xv * B[j] + K[j]);
    }
    *(bf16x8*)&dx[m * C + c0] = o;
  }
}

// ---------------------------------------------------------------------------
// Pooling (NHWC): max with argmax (u8), avg, global-avg; backwards.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void maxpool_fwd(
    const bf16* __restrict__ x, bf16* __restrict__ y,
    unsigned char* __restrict__ arg, int Nb, int H, int W, int C, int P,
    int Q, int R, int S, int sh, int sw, int ph, int pw) {
  long total = (long)Nb * P * Q * (C / 8);
  int cv = C / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c0 = (int)(i % cv) * 8;
    long npq = i / cv;
    int q = (int)(npq % Q);
    int p = (int)((npq / Q) % P);
    int n = (int)(npq / ((long)P * Q));
    float best[8];
    int bidx[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      best[j] = -3.0e38f;
      bidx[j] = 0;
    }
    for (int r = 0; r < R; ++r) {
      int ih = p * sh - ph + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < S; ++s) {
        int iw = q * sw - pw + s;
        if (iw < 0 || iw >= W) continue;
        bf16x8 v =
            *(const bf16x8*)&x[(((long)n * H + ih) * W + iw) * C + c0];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf2f(v[j]);
          if (f > best[j]) {
            best[j] = f;
            bidx[j] = r * S + s;
          }
        }
      }
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      o[j] = f2bf(best[j]);
      arg[(npq * C + c0) + j] = (unsigned char)bidx[j];
    }
    *(bf16x8*)&y[npq * C + c0] = o;
  }
}

// gather form: each INPUT position sums the windows whose argmax points
// at it (no atomics, no zero-fill, bf16 out directly)
__global__ __launch_bounds__(256) void maxpool_bwd(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ arg,
    bf16* __restrict__ dx, int Nb, int H, int W, int C, int P, int Q,
    int R, int S, int sh, int sw, int ph, int pw) {
  // gather form, 8 channels per thread (no atomics): for each input
  // pixel, scan the few output windows covering it and claim the grads
  // whose argmax points here.
  long total = (long)Nb * H * W * (C / 8);
  int cv = C / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c0 = (int)(i % cv) * 8;
    long nhw = i / cv;
    int w_ = (int)(nhw % W);
    int h = (int)((nhw / W) % H);
    int n = (int)(nhw / ((long)H * W));
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = 0.f;
    int p_lo = (h + ph - R + sh) >= 0 ? (h + ph - R + sh) / sh : 0;
    int p_hi = min((h + ph) / sh, P - 1);
    int q_lo = (w_ + pw - S + sw) >= 0 ? (w_ + pw - S + sw) / sw : 0;
    int q_hi = min((w_ + pw) / sw, Q - 1);
    for (int p = p_lo; p <= p_hi; ++p) {
      int r = h + ph - p * sh;
      if (r < 0 || r >= R) continue;
      for (int q = q_lo; q <= q_hi; ++q) {
        int sx = w_ + pw - q * sw;
        if (sx < 0 || sx >= S) continue;
        long oi = (((long)n * P + p) * Q + q) * C + c0;
        unsigned char want = (unsigned char)(r * S + sx);
        u64 a8 = *(const u64*)&arg[oi];
        bf16x8 g8 = *(const bf16x8*)&dy[oi];
        const unsigned char* ab = (const unsigned char*)&a8;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (ab[j] == want) acc[j] += bf2f(g8[j]);
      }
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(acc[j]);
    *(bf16x8*)&dx[(((long)n * H + h) * W + w_) * C + c0] = o;
  }
}

// Materialized im2col for the small-C stem conv (7x7 C=3): writes the
// padded [N*P*Q, kdim_pad] GEMM operand directly (k order (r, s, c),
// zeros in the %64 pad tail and outside the image).  One block per
// output row (n, p): the R input rows are staged once through LDS
// (x read once per (n,p) instead of once per output element) and the
// whole [Q][kpadded] slab is emitted with bf16x8 stores.
// Table-driven gather: the k -> (r, s, c) unpack is FIXED across every
// output row, so it is computed once per block into an LDS offset table
// (the per-element %C / /C / %S divisions made the first version
// VALU-bound at ~19x the HBM roofline).  The staged rows carry zeroed
// left/right column pads sized for the filter overhang, and OOB input
// rows stage as zeros, so the inner loop is branch-free: one LDS read
// per output element at a precomputed offset.
__global__ __launch_bounds__(256) void im2col_stem(
    const bf16* __restrict__ x, bf16* __restrict__ out, int Nb, int H,
    int W, int C, int P, int Q, int R, int S, int sh, int sw, int ph,
    int pw, int dh, int dw, int kdim, int kpadded, int lpad, int wpc) {
  extern __shared__ bf16 xrow[];  // [R][wpc] rows, then int koff[kpadded]
  int* koff = (int*)&xrow[(long)R * wpc];
  const int WC = W * C;
  for (int k = threadIdx.x; k < kpadded; k += 256) {
    int c = k % C, rs = k / C;
    int sx = rs % S, r = rs / S;
    koff[k] = r * wpc + sx * dw * C + c + lpad;
  }
  // zero the column pads once; staging only ever writes the interior
  for (int i = threadIdx.x; i < R * wpc; i += 256) {
    int col = i % wpc;
    if (col < lpad || col >= lpad + WC) xrow[i] = (bf16)0;
  }
  const int wcw = (WC + 1) / 2;  // u32 words per interior row
  const int kv = kpadded / 8;
  __syncthreads();
  for (long np = blockIdx.x; np < (long)Nb * P; np += gridDim.x) {
    int p = (int)(np % P);
    int n = (int)(np / P);
    int ih0 = p * sh - ph;
    const bf16* xn = x + (long)n * H * WC;
    for (int idx = threadIdx.x; idx < R * wcw; idx += 256) {
      int r = idx / wcw, wd = idx % wcw;
      int ih = ih0 + r * dh;
      u32 v = 0;
      if (ih >= 0 && ih < H) {
        const bf16* rowp = xn + (long)ih * WC;
        if (wd * 2 + 1 < WC) {
          v = *(const u32*)(rowp + wd * 2);
        } else {  // odd-WC tail: avoid reading past the row
          u32 lo = *(const unsigned short*)(rowp + wd * 2);
          v = lo;
        }
      }
      // lpad is even -> 4-byte aligned u32 store
      *(u32*)&xrow[r * wpc + lpad + wd * 2] = v;
    }
    __syncthreads();
    // emit [Q][kpadded]
    bf16* orow = out + np * (long)Q * kpadded;
    for (int i = threadIdx.x; i < Q * kv; i += 256) {
      int q = i / kv;
      int j0 = (i % kv) * 8;
      int base = (q * sw - pw) * C;
      bf16x8 o;
      if (j0 + 8 <= kdim) {
#pragma unroll
        for (int e = 0; e < 8; ++e) o[e] = xrow[koff[j0 + e] + base];
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          int k = j0 + e;
          o[e] = k < kdim ? xrow[koff[k] + base] : (bf16)0;
        }
      }
      *(bf16x8*)&orow[(long)q * kpadded + j0] = o;
    }
    __syncthreads();
  }
}

template <bool COUNT_PAD>
__global__ __launch_bounds__(256) void avgpool_fwd(
    const bf16* __restrict__ x, bf16* __restrict__ y, int Nb, int H, int W,
    int C, int P, int Q, int R, int S, int sh, int sw, int ph, int pw) {
  long total = (long)Nb * P * Q * (C / 8);
  int cv = C / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c0 = (int)(i % cv) * 8;
    long npq = i / cv;
    int q = (int)(npq % Q);
    int p = (int)((npq / Q) % P);
    int n = (int)(npq / ((long)P * Q));
    float acc[8] = {0};
    int cnt = 0;
    for (int r = 0; r < R; ++r) {
      int ih = p * sh - ph + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < S; ++s) {
        int iw = q * sw - pw + s;
        if (iw < 0 || iw >= W) continue;
        bf16x8 v =
            *(const bf16x8*)&x[(((long)n * H + ih) * W + iw) * C + c0];
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf2f(v[j]);
        ++cnt;
      }
    }
    float div = COUNT_PAD ? (float)(R * S) : (float)max(cnt, 1);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(acc[j] / div);
    *(bf16x8*)&y[npq * C + c0] = o;
  }
}

// global average pool over H*W: y[n, c] (+ backward broadcast)
__global__ __launch_bounds__(256) void gap_fwd(const bf16* __restrict__ x,
                                               bf16* __restrict__ y, int Nb,
                                               long HW, int C) {
  // one block per n, threads cover channels in chunks of 8
  int n = blockIdx.x;
  for (int c0 = threadIdx.x * 8; c0 < C; c0 += blockDim.x * 8) {
    float acc[8] = {0};
    const bf16* base = x + (long)n * HW * C + c0;
    for (long m = 0; m < HW; ++m) {
      bf16x8 v = *(const bf16x8*)&base[m * C];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += bf2f(v[j]);
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(acc[j] / (float)HW);
    *(bf16x8*)&y[(long)n * C + c0] = o;
  }
}

__global__ __launch_bounds__(256) void gap_bwd(const bf16* __restrict__ dy,
                                               bf16* __restrict__ dx, int Nb,
                                               long HW, int C) {
  long total = (long)Nb * HW * (C / 8);
  int cv = C / 8;
  float inv = 1.f / (float)HW;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c0 = (int)(i % cv) * 8;
    long nm = i / cv;
    int n = (int)(nm / HW);
    bf16x8 g = *(const bf16x8*)&dy[(long)n * C + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(g[j]) * inv);
    *(bf16x8*)&dx[i * 8] = o;
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

int bn_gy(long M, int C) {
  // fill the chip: ~768 blocks total regardless of C (narrow-C layers
  // otherwise leave most CUs idle in the reduce)
  int cb = cdiv_h(C, 64);
  long gy = min((M + 255) / 256, (long)(768 / cb + 1));
  return (int)(gy < 1 ? 1 : gy);
}

static dim3 bn_apply_grid(long M, int C) {
  int cb = cdiv_h(C, 64);
  long gy = (M + 31) / 32;
  long cap = 2048 / cb + 1;
  if (gy > cap) gy = cap;
  if (gy < 1) gy = 1;
  return dim3(cb, (unsigned)gy);
}

void bn_fwd_train(const bf16* x, bf16* y, float* psum, float* psumsq,
                  float* save_mean, float* save_rstd, float* running_mean,
                  float* running_var, const float* gamma, const float* beta,
                  long M, int C, float eps, float momentum, bool relu,
                  const bf16* resid, hipStream_t s, const float* ext_sums) {
  int gy = bn_gy(M, C);
  if (ext_sums) {
    // per-channel sum / sum-of-squares were already accumulated by the
    // producing conv's GEMM epilogue (gemm_tn.hip fused-stats path) in
    // STATS_SLOTS slot copies: fold + finalize, skipping the
    // bn_stats_partial re-read of x entirely.
    bn_finalize<<<cdiv_h(C, 256), 256, 0, s>>>(
        ext_sums, ext_sums + C, save_mean, save_rstd, running_mean,
        running_var, M, C, gy, eps, momentum, 1, STATS_SLOTS, 2L * C);
  } else {
    // psum/psumsq are one contiguous [2,C] buffer: single fill
    HIP_CHECK(hipMemsetAsync(psum, 0, 2L * C * sizeof(float), s));
    int cb = cdiv_h(C, 64);
    dim3 g1(cb, gy);
    bn_stats_partial<<<g1, 256, 0, s>>>(x, psum, psumsq, M, C);
    bn_finalize<<<cdiv_h(C, 256), 256, 0, s>>>(
        psum, psumsq, save_mean, save_rstd, running_mean, running_var, M, C,
        gy, eps, momentum, 1, 1, 0);
  }
  dim3 g2 = bn_apply_grid(M, C);
  if (resid)
    bn_apply_add<<<g2, 256, 0, s>>>(x, resid, y, save_mean, save_rstd,
                                    gamma, beta, M, C);
  else if (relu)
    bn_apply<true><<<g2, 256, 0, s>>>(x, y, save_mean, save_rstd, gamma,
                                      beta, M, C);
  else
    bn_apply<false><<<g2, 256, 0, s>>>(x, y, save_mean, save_rstd, gamma,
                                       beta, M, C);
}

void bn_fwd_infer(const bf16* x, bf16* y, const float* rmean,
                  const float* rvar, const float* gamma, const float* beta,
                  long M, int C, float eps, bool relu, const bf16* resid,
                  hipStream_t s) {
  dim3 g = bn_apply_grid(M, C);
  if (resid)
    bn_infer_add<<<g, 256, 0, s>>>(x, resid, y, rmean, rvar, gamma, beta,
                                   M, C, eps);
  else if (relu)
    bn_infer<true><<<g, 256, 0, s>>>(x, y, rmean, rvar, gamma, beta, M, C,
                                     eps);
  else
    bn_infer<false><<<g, 256, 0, s>>>(x, y, rmean, rvar, gamma, beta, M, C,
                                      eps);
}

void bn_bwd(const bf16* x, const bf16* dy, bf16* dx, const float* save_mean,
            const float* save_rstd, const float* gamma, const float* beta,
            float* dgamma, float* dbeta, long M, int C, bool relu,
            hipStream_t s) {
  // dgamma/dbeta are one contiguous [2,C] buffer: single fill
  HIP_CHECK(hipMemsetAsync(dbeta < dgamma ? dbeta : dgamma, 0,
                           2L * C * sizeof(float), s));
  int cb = cdiv_h(C, 64);
  int gy = bn_gy(M, C);
  dim3 g1(cb, gy);
  dim3 g2 = bn_apply_grid(M, C);
  if (relu)
    bn_bwd_reduce<true><<<g1, 256, 0, s>>>(x, dy, save_mean, save_rstd,
                                           gamma, beta, dbeta, dgamma, M, C);
  else
    bn_bwd_reduce<false><<<g1, 256, 0, s>>>(x, dy, save_mean, save_rstd,
                                            gamma, beta, dbeta, dgamma, M,
                                            C);
  if (relu)
    bn_bwd_apply<true><<<g2, 256, 0, s>>>(x, dy, dx, save_mean, save_rstd,
                                          gamma, beta, dbeta, dgamma, M, C);
  else
    bn_bwd_apply<false><<<g2, 256, 0, s>>>(x, dy, dx, save_mean, save_rstd,
                                           gamma, beta, dbeta, dgamma, M, C);
}

void maxpool_nhwc(const bf16* x, bf16* y, unsigned char* arg, int Nb, int H,
                  int W, int C, int P, int Q, int R, int S, int sh, int sw,
                  int ph, int pw, hipStream_t s) {
  dim3 g = ew_grid((long)Nb * P * Q * (C / 8));
  maxpool_fwd<<<g, 256, 0, s>>>(x, y, arg, Nb, H, W, C, P, Q, R, S, sh, sw,
                                ph, pw);
}

void im2col_stem_bf16(const bf16* x, bf16* out, int Nb, int H, int W,
                      int C, int P, int Q, int R, int S, int sh, int sw,
                      int ph, int pw, int dh, int dw, int kdim, int kpadded,
                      hipStream_t s) {
  long blocks = (long)Nb * P;
  if (blocks > 16384) blocks = 16384;
  // padded row image: left pad covers the filter's left overhang (pw),
  // right pad its right overhang; lpad even keeps u32 staging aligned
  int lpad = pw * C;
  lpad += lpad & 1;
  int iw_max = (Q - 1) * sw - pw + (S - 1) * dw;
  int rext = iw_max >= W ? (iw_max - W + 1) * C : 0;
  int wpc = lpad + W * C + rext;
  wpc += wpc & 1;
  int lds = R * wpc * sizeof(bf16) + kpadded * sizeof(int);
  im2col_stem<<<dim3((unsigned)blocks), 256, lds, s>>>(
      x, out, Nb, H, W, C, P, Q, R, S, sh, sw, ph, pw, dh, dw, kdim,
      kpadded, lpad, wpc);
}

void maxpool_nhwc_bwd(const bf16* dy, const unsigned char* arg, bf16* dx,
                      int Nb, int H, int W, int C, int P, int Q, int R,
                      int S, int sh, int sw, int ph, int pw, hipStream_t s) {
  dim3 g = ew_grid((long)Nb * H * W * (C / 8), 256, 8192);
  maxpool_bwd<<<g, 256, 0, s>>>(dy, arg, dx, Nb, H, W, C, P, Q, R, S, sh,
                                sw, ph, pw);
}

void avgpool_nhwc(const bf16* x, bf16* y, int Nb, int H, int W, int C,
                  int P, int Q, int R, int S, int sh, int sw, int ph,
                  int pw, bool count_pad, hipStream_t s) {
  dim3 g = ew_grid((long)Nb * P * Q * (C / 8));
  if (count_pad)
    avgpool_fwd<true><<<g, 256, 0, s>>>(x, y, Nb, H, W, C, P, Q, R, S, sh,
                                        sw, ph, pw);
  else
    avgpool_fwd<false><<<g, 256, 0, s>>>(x, y, Nb, H, W, C, P, Q, R, S, sh,
                                         sw, ph, pw);
}

void gap_nhwc(const bf16* x, bf16* y, int Nb, long HW, int C,
              hipStream_t s) {
  gap_fwd<<<Nb, 256, 0, s>>>(x, y, Nb, HW, C);
}

void gap_nhwc_bwd(const bf16* dy, bf16* dx, int Nb, long HW, int C,
                  hipStream_t s) {
  dim3 g = ew_grid((long)Nb * HW * (C / 8));
  gap_bwd<<<g, 256, 0, s>>>(dy, dx, Nb, HW, C);
}


// ---------------------------------------------------------------------------
// Depthwise conv (groups == C), NHWC bf16: channels are the fast dim, so
// each thread owns 8 channels of one output pixel and walks the RxS
// window (reference parity: depthwise_convolution.cu).  Memory-bound.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void dwconv_fwd(
    const bf16* __restrict__ x, const bf16* __restrict__ w,  // w [R,S,C]
    const float* __restrict__ bias, bf16* __restrict__ y, int Nb, int H,
    int W, int C, int P, int Q, int R, int S, int sh, int sw, int ph,
    int pw, int dh, int dw_) {
  int cv = C / 8;
  long total = (long)Nb * P * Q * cv;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c0 = (int)(i % cv) * 8;
    long npq = i / cv;
    int q = (int)(npq % Q);
    int p = (int)((npq / Q) % P);
    int n = (int)(npq / ((long)P * Q));
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = bias ? bias[c0 + j] : 0.f;
    for (int r = 0; r < R; ++r) {
      int ih = p * sh - ph + r * dh;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < S; ++s) {
        int iw = q * sw - pw + s * dw_;
        if (iw < 0 || iw >= W) continue;
        bf16x8 xv = *(const bf16x8*)&x[(((long)n * H + ih) * W + iw) * C
                                       + c0];
        bf16x8 wv = *(const bf16x8*)&w[((long)r * S + s) * C + c0];
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf2f(xv[j]) * bf2f(wv[j]);
      }
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(acc[j]);
    *(bf16x8*)&y[npq * C + c0] = o;
  }
}

// dgrad: gather form over covering output windows (cf. maxpool_bwd)
__global__ __launch_bounds__(256) void dwconv_dgrad(
    const bf16* __restrict__ dy, const bf16* __restrict__ w,
    bf16* __restrict__ dx, int Nb, int H, int W, int C, int P, int Q,
    int R, int S, int sh, int sw, int ph, int pw) {
  int cv = C / 8;
  long total = (long)Nb * H * W * cv;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c0 = (int)(i % cv) * 8;
    long nhw = i / cv;
    int w_ = (int)(nhw % W);
    int h = (int)((nhw / W) % H);
    int n = (int)(nhw / ((long)H * W));
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = 0.f;
    int p_lo = (h + ph - R + sh) >= 0 ? (h + ph - R + sh) / sh : 0;
    int p_hi = min((h + ph) / sh, P - 1);
    int q_lo = (w_ + pw - S + sw) >= 0 ? (w_ + pw - S + sw) / sw : 0;
    int q_hi = min((w_ + pw) / sw, Q - 1);
    for (int p = p_lo; p <= p_hi; ++p) {
      int r = h + ph - p * sh;
      if (r < 0 || r >= R) continue;
      for (int q = q_lo; q <= q_hi; ++q) {
        int sx = w_ + pw - q * sw;
        if (sx < 0 || sx >= S) continue;
        bf16x8 g = *(const bf16x8*)&dy[(((long)n * P + p) * Q + q) * C
                                       + c0];
        bf16x8 wv = *(const bf16x8*)&w[((long)r * S + sx) * C + c0];
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf2f(g[j]) * bf2f(wv[j]);
      }
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(acc[j]);
    *(bf16x8*)&dx[(((long)n * H + h) * W + w_) * C + c0] = o;
  }
}

// wgrad: per-(r,s,c) reduce over (n,p,q); stats-partial geometry + atomics
__global__ __launch_bounds__(256) void dwconv_wgrad(
    const bf16* __restrict__ x, const bf16* __restrict__ dy,
    float* __restrict__ dwf, int Nb, int H, int W, int C, int P, int Q,
    int R, int S, int sh, int sw, int ph, int pw) {
  // block: 256 thr = 32 row-lanes x 8 chunks; grid.x = rs * C/64 blocks
  const int nc64 = (C + 63) / 64;
  const int rs = blockIdx.x / nc64;
  const int cblk = (blockIdx.x % nc64) * 64;
  const int r = rs / S, s = rs % S;
  const int t = threadIdx.x;
  const int chunk = t & 7, rlane = t >> 3;
  const int c0 = cblk + chunk * 8;
  const bool active = c0 + 8 <= C;
  long M = (long)Nb * P * Q;
  float acc[8] = {0};
  for (long m = blockIdx.y * 32 + rlane; active && m < M;
       m += 32L * gridDim.y) {
    int q = (int)(m % Q);
    int p = (int)((m / Q) % P);
    int n = (int)(m / ((long)P * Q));
    int ih = p * sh - ph + r;
    int iw = q * sw - pw + s;
    if (ih < 0 || ih >= H || iw < 0 || iw >= W) continue;
    bf16x8 xv = *(const bf16x8*)&x[(((long)n * H + ih) * W + iw) * C + c0];
    bf16x8 g = *(const bf16x8*)&dy[m * C + c0];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += bf2f(xv[j]) * bf2f(g[j]);
  }
  __shared__ float red[32][64];
#pragma unroll
  for (int j = 0; j < 8; ++j) red[rlane][chunk * 8 + j] = acc[j];
  __syncthreads();
  if (t < 64) {
    float f = 0;
#pragma unroll
    for (int rr = 0; rr < 32; ++rr) f += red[rr][t];
    int c = cblk + t;
    if (c < C) atomicAdd(&dwf[(long)rs * C + c], f);
  }
}

void dwconv_fwd_bf16(const bf16* x, const bf16* w, const float* bias,
                     bf16* y, int Nb, int H, int W, int C, int P, int Q,
                     int R, int S, int sh, int sw, int ph, int pw, int dh,
                     int dw_, hipStream_t s) {
  dim3 g = ew_grid((long)Nb * P * Q * (C / 8), 256, 8192);
  dwconv_fwd<<<g, 256, 0, s>>>(x, w, bias, y, Nb, H, W, C, P, Q, R, S, sh,
                               sw, ph, pw, dh, dw_);
}

void dwconv_dgrad_bf16(const bf16* dy, const bf16* w, bf16* dx, int Nb,
                       int H, int W, int C, int P, int Q, int R, int S,
                       int sh, int sw, int ph, int pw, hipStream_t s) {
  dim3 g = ew_grid((long)Nb * H * W * (C / 8), 256, 8192);
  dwconv_dgrad<<<g, 256, 0, s>>>(dy, w, dx, Nb, H, W, C, P, Q, R, S, sh,
                                 sw, ph, pw);
}

void dwconv_wgrad_bf16(const bf16* x, const bf16* dy, float* dwf, int Nb,
                       int H, int W, int C, int P, int Q, int R, int S,
                       int sh, int sw, int ph, int pw, hipStream_t s) {
  int nc64 = (C + 63) / 64;
  long M = (long)Nb * P * Q;
  long gy = min((M + 31) / 32, (long)(1024 / (R * S * nc64) + 1));
  if (gy < 1) gy = 1;
  HIP_CHECK(hipMemsetAsync(dwf, 0, sizeof(float) * R * S * C, s));
  dwconv_wgrad<<<dim3(R * S * nc64, (unsigned)gy), 256, 0, s>>>(
      x, dy, dwf, Nb, H, W, C, P, Q, R, S, sh, sw, ph, pw);
}

}  // namespace mxa
