// Dropout (counter-based RNG), Embedding gather/scatter-grad, and the
// generic vectorized elementwise/activation family for gfx950.
// Reference parity: src/operator/nn/dropout.cu (philox mask+scale fused),
// tensor/indexing_op.cu (Embedding/AddTakeGrad), mxnet_op.h Kernel::Launch
// (the generic elementwise launcher — here a bf16x8 grid-stride template).
#include "common.h"

namespace mxa {

// ---------------------------------------------------------------------------
// Dropout: philox-style counter hash -> mask byte + scaled output, fused.
// ---------------------------------------------------------------------------

// seed = *seed_ptr + salt: the device seed word evolves via rng_bump
// (graph-safe — a captured step replays the bump, so dropout masks differ
// per replay), the per-call salt decorrelates call sites within a step.
__global__ __launch_bounds__(256) void dropout_fwd(
    const bf16* __restrict__ x, bf16* __restrict__ y,
    unsigned char* __restrict__ mask, long nvec,
    const u64* __restrict__ seed_ptr, u64 salt, float p, float scale) {
  // finalize (seed+salt) BEFORE mixing with the element index: the raw
  // sum is linear in the step counter, and hash(seed + c*s + c*(i+1))
  // would make step s+1's mask a one-element SHIFT of step s's —
  // perfectly correlated dropout across steps.
  u64 seed = seed_finalize(*seed_ptr + salt);
  u32 thresh = (u32)(p * 4294967296.0f);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 v = *(const bf16x8*)&x[i * 8];
    bf16x8 o;
    unsigned char mb[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      u32 r = hash_u32(seed, i * 8 + j);
      bool keep = r >= thresh;
      mb[j] = keep;
      o[j] = keep ? f2bf(bf2f(v[j]) * scale) : (bf16)0.f;
    }
    *(bf16x8*)&y[i * 8] = o;
    *(u64*)&mask[i * 8] = *(const u64*)mb;
  }
}

__global__ __launch_bounds__(256) void dropout_bwd(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ mask,
    bf16* __restrict__ dx, long nvec, float scale) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 g = *(const bf16x8*)&dy[i * 8];
    u64 m8 = *(const u64*)&mask[i * 8];
    const unsigned char* mb = (const unsigned char*)&m8;
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = mb[j] ? f2bf(bf2f(g[j]) * scale) : (bf16)0.f;
    *(bf16x8*)&dx[i * 8] = o;
  }
}

// ---------------------------------------------------------------------------
// Embedding: row gather fwd; scatter-add (atomic fp32) bwd.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void embedding_fwd(
    const long* __restrict__ idx, const bf16* __restrict__ W,
    bf16* __restrict__ out, long ntok, int D, long vocab) {
  long total = ntok * (D / 8);
  int dv = D / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long tok = i / dv;
    int d0 = (int)(i % dv) * 8;
    long row = idx[tok];
    if (row < 0 || row >= vocab) {
      *(bf16x8*)&out[tok * D + d0] = bf16x8{};
      continue;
    }
    *(bf16x8*)&out[tok * D + d0] = *(const bf16x8*)&W[row * D + d0];
  }
}

__global__ __launch_bounds__(256) void embedding_bwd(
    const long* __restrict__ idx, const bf16* __restrict__ dy,
    float* __restrict__ dW, long ntok, int D, long vocab) {
  long total = ntok * D;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long tok = i / D;
    int d = (int)(i % D);
    long row = idx[tok];
    if (row < 0 || row >= vocab) continue;
    atomicAdd(&dW[row * D + d], bf2f(dy[i]));
  }
}

// ---------------------------------------------------------------------------
// Generic elementwise activations (bf16x8, fused bwd forms).
// op ids: 0 relu, 1 sigmoid, 2 tanh, 3 gelu(erf), 4 silu, 5 gelu(tanh)
// ---------------------------------------------------------------------------

DEVINL float act_fwd_one(int OP, float x) {
  switch (OP) {
    case 0: return x > 0 ? x : 0;
    case 1: return 1.f / (1.f + __expf(-x));
    case 2: return tanhf(x);
    case 3: return 0.5f * x * (1.f + erff(x * 0.70710678f));
    case 4: return x / (1.f + __expf(-x));
    default: {
      float c = 0.7978845608f * (x + 0.044715f * x * x * x);
      return 0.5f * x * (1.f + tanhf(c));
    }
  }
}

DEVINL float act_bwd_one(int OP, float x, float g) {
  switch (OP) {
    case 0: return x > 0 ? g : 0;
    case 1: {
      float s = 1.f / (1.f + __expf(-x));
      return g * s * (1.f - s);
    }
    case 2: {
      float t = tanhf(x);
      return g * (1.f - t * t);
    }
    case 3: {
      float cdf = 0.5f * (1.f + erff(x * 0.70710678f));
      float pdf = 0.3989422804f * __expf(-0.5f * x * x);
      return g * (cdf + x * pdf);
    }
    case 4: {
      float s = 1.f / (1.f + __expf(-x));
      return g * s * (1.f + x * (1.f - s));
    }
    default: {
      float c = 0.7978845608f * (x + 0.044715f * x * x * x);
      float t = tanhf(c);
      float dc = 0.7978845608f * (1.f + 3.f * 0.044715f * x * x);
      return g * (0.5f * (1.f + t) + 0.5f * x * (1.f - t * t) * dc);
    }
  }
}

template <int OP>
__global__ __launch_bounds__(256) void act_fwd(const bf16* __restrict__ x,
                                               bf16* __restrict__ y,
                                               long nvec) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 v = *(const bf16x8*)&x[i * 8];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(act_fwd_one(OP, bf2f(v[j])));
    *(bf16x8*)&y[i * 8] = o;
  }
}

template <int OP>
__global__ __launch_bounds__(256) void act_bwd(const bf16* __restrict__ x,
                                               const bf16* __restrict__ dy,
                                               bf16* __restrict__ dx,
                                               long nvec) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 v = *(const bf16x8*)&x[i * 8];
    bf16x8 g = *(const bf16x8*)&dy[i * 8];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bf(act_bwd_one(OP, bf2f(v[j]), bf2f(g[j])));
    *(bf16x8*)&dx[i * 8] = o;
  }
}

// bf16 <-> f32 cast + scaled copy (AMP plumbing)
__global__ __launch_bounds__(256) void scale_cast_f32_bf16(
    const float* __restrict__ x, bf16* __restrict__ y, long nvec,
    float scale) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    f32x4 v = *(const f32x4*)&x[i * 4];
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = f2bf(v[j] * scale);
    *(bf16x4*)&y[i * 4] = o;
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

// native sampling (reference src/operator/random/sample_op.* — philox-
// style counter hash, same generator family as dropout)
__global__ __launch_bounds__(256) void rng_uniform_kernel(
    bf16* __restrict__ y, long nvec, const u64* __restrict__ seed_ptr,
    u64 salt, float lo, float span) {
  u64 seed = seed_finalize(*seed_ptr + salt);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      u32 r = hash_u32(seed, i * 8 + j);
      o[j] = f2bf(lo + span * (float)r * 2.3283064365386963e-10f);
    }
    *(bf16x8*)&y[i * 8] = o;
  }
}

__global__ __launch_bounds__(256) void rng_normal_kernel(
    bf16* __restrict__ y, long nvec, const u64* __restrict__ seed_ptr,
    u64 salt, float mu, float sigma) {
  u64 seed = seed_finalize(*seed_ptr + salt);
  const float TWO_PI = 6.28318530718f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; j += 2) {
      u32 r1 = hash_u32(seed, (i * 8 + j) * 2);
      u32 r2 = hash_u32(seed, (i * 8 + j) * 2 + 1);
      float u1 = ((float)r1 + 1.0f) * 2.3283064365386963e-10f;
      float u2 = (float)r2 * 2.3283064365386963e-10f;
      float rad = sqrtf(-2.0f * __logf(u1)) * sigma;
      o[j] = f2bf(mu + rad * __cosf(TWO_PI * u2));
      o[j + 1] = f2bf(mu + rad * __sinf(TWO_PI * u2));
    }
    *(bf16x8*)&y[i * 8] = o;
  }
}

void rng_uniform_bf16(bf16* y, long n, const u64* seed, u64 salt, float lo,
                      float hi, hipStream_t s) {
  rng_uniform_kernel<<<ew_grid(n / 8), 256, 0, s>>>(y, n / 8, seed, salt,
                                                    lo, hi - lo);
}

void rng_normal_bf16(bf16* y, long n, const u64* seed, u64 salt, float mu,
                     float sigma, hipStream_t s) {
  rng_normal_kernel<<<ew_grid(n / 8), 256, 0, s>>>(y, n / 8, seed, salt,
                                                   mu, sigma);
}

__global__ void rng_bump(u64* s) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *s += 0x9e3779b97f4a7c15ULL;
}

void dropout_fwd_bf16(const bf16* x, bf16* y, unsigned char* mask, long n,
                      const u64* seed_ptr, u64 salt, float p,
                      hipStream_t s) {
  long nvec = n / 8;
  dropout_fwd<<<ew_grid(nvec), 256, 0, s>>>(x, y, mask, nvec, seed_ptr,
                                            salt, p, 1.f / (1.f - p));
}

void rng_bump_launch(u64* seed_ptr, hipStream_t s) {
  rng_bump<<<1, 1, 0, s>>>(seed_ptr);
}

void dropout_bwd_bf16(const bf16* dy, const unsigned char* mask, bf16* dx,
                      long n, float p, hipStream_t s) {
  long nvec = n / 8;
  dropout_bwd<<<ew_grid(nvec), 256, 0, s>>>(dy, mask, dx, nvec,
                                            1.f / (1.f - p));
}

void embedding_fwd_bf16(const long* idx, const bf16* W, bf16* out,
                        long ntok, int D, long vocab, hipStream_t s) {
  embedding_fwd<<<ew_grid(ntok * (D / 8)), 256, 0, s>>>(idx, W, out, ntok,
                                                        D, vocab);
}

void embedding_bwd_bf16(const long* idx, const bf16* dy, float* dW,
                        long ntok, int D, long vocab, hipStream_t s) {
  HIP_CHECK(hipMemsetAsync(dW, 0, vocab * D * sizeof(float), s));
  embedding_bwd<<<ew_grid(ntok * D), 256, 0, s>>>(idx, dy, dW, ntok, D,
                                                  vocab);
}

// Column sum (bias gradient): out[c] = sum_m x[m][c], bf16 in / f32 out.
// One thread per column, rows split over blockIdx.y, f32 atomics to finish
// (replaces torch's materialized .float() copy + reduce_kernel pair).
// 256 thr = 32 row-lanes x 8 chunks of 8 channels (same geometry as
// bn_stats_partial): bf16x8 row loads, LDS cross-lane reduce, one atomic
// per channel per block — the 1-thread-per-column form was latency-bound
// on the small-N bias grads.
__global__ __launch_bounds__(256) void colsum_kernel(
    const bf16* __restrict__ x, float* __restrict__ out, long m, int N) {
  const int cblk = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int chunk = t & 7, rlane = t >> 3;
  const int c0 = cblk + chunk * 8;
  const bool active = c0 + 8 <= N;
  float acc[8] = {0};
  for (long r = blockIdx.y * 32 + rlane; active && r < m;
       r += 32L * gridDim.y) {
    bf16x8 v = *(const bf16x8*)&x[r * N + c0];
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[i] += bf2f(v[i]);
  }
  __shared__ float red[32][64];
#pragma unroll
  for (int i = 0; i < 8; ++i) red[rlane][chunk * 8 + i] = acc[i];
  __syncthreads();
  if (t < 64) {
    float f = 0;
#pragma unroll
    for (int r = 0; r < 32; ++r) f += red[r][t];
    int c = cblk + t;
    if (c < N) {
      if (gridDim.y > 1)
        atomicAdd(&out[c], f);
      else
        out[c] = f;
    }
  }
}

// tail columns (N % 8 != 0 can't take bf16x8 loads): scalar finish
__global__ void colsum_tail(const bf16* __restrict__ x,
                            float* __restrict__ out, long m, int N,
                            int c0) {
  int c = c0 + blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= N) return;
  float acc = 0.f;
  for (long r = 0; r < m; ++r) acc += bf2f(x[r * N + c]);
  out[c] = acc;
}

void colsum_bf16(const bf16* x, float* out, long m, int N, hipStream_t s) {
  if (N % 8 != 0) {
    // rows not 16-B aligned -> scalar kernel over all columns
    colsum_tail<<<cdiv_h(N, 256), min(N, 256), 0, s>>>(x, out, m, N, 0);
    return;
  }
  int cb = cdiv_h(N, 64);
  long gy = min((m + 31) / 32, (long)(1024 / cb + 1));
  if (gy < 1) gy = 1;
  if (gy > 1) HIP_CHECK(hipMemsetAsync(out, 0, N * sizeof(float), s));
  colsum_kernel<<<dim3(cb, (unsigned)gy), 256, 0, s>>>(x, out, m, N);
}

void act_fwd_bf16(int op, const bf16* x, bf16* y, long n, hipStream_t s) {
  long nvec = n / 8;
  dim3 g = ew_grid(nvec);
  switch (op) {
    case 0: act_fwd<0><<<g, 256, 0, s>>>(x, y, nvec); break;
    case 1: act_fwd<1><<<g, 256, 0, s>>>(x, y, nvec); break;
    case 2: act_fwd<2><<<g, 256, 0, s>>>(x, y, nvec); break;
    case 3: act_fwd<3><<<g, 256, 0, s>>>(x, y, nvec); break;
    case 4: act_fwd<4><<<g, 256, 0, s>>>(x, y, nvec); break;
    default: act_fwd<5><<<g, 256, 0, s>>>(x, y, nvec); break;
  }
}

void act_bwd_bf16(int op, const bf16* x, const bf16* dy, bf16* dx, long n,
                  hipStream_t s) {
  long nvec = n / 8;
  dim3 g = ew_grid(nvec);
  switch (op) {
    case 0: act_bwd<0><<<g, 256, 0, s>>>(x, dy, dx, nvec); break;
    case 1: act_bwd<1><<<g, 256, 0, s>>>(x, dy, dx, nvec); break;
    case 2: act_bwd<2><<<g, 256, 0, s>>>(x, dy, dx, nvec); break;
    case 3: act_bwd<3><<<g, 256, 0, s>>>(x, dy, dx, nvec); break;
    case 4: act_bwd<4><<<g, 256, 0, s>>>(x, dy, dx, nvec); break;
    default: act_bwd<5><<<g, 256, 0, s>>>(x, dy, dx, nvec); break;
  }
}

}  // namespace mxa
