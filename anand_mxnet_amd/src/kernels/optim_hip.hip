#include "hip/hip_runtime.h"
// Multi-tensor fused optimizer updates for gfx950.
// Reference parity: src/operator/optimizer_op.cu — sgd_mom_update,
// mp_sgd_mom_update (fp32 master weights for bf16 params), adam_update,
// multi_* aggregated variants, multi_sum_sq.
//
// Design: one launch per parameter GROUP.  The host packs per-tensor
// {pointers, numel} into a device-side descriptor array; blocks walk a flat
// chunk list (chunk = 65536 elements) so one kernel covers every tensor.
// bf16 weights/grads are vectorized 8-wide; all math accumulates fp32.
#include "common.h"

namespace mxa {

struct TensorDesc {
  void* w;      // bf16 or f32 weight
  void* g;      // grad (same dtype as w)
  float* m;     // momentum / adam m
  float* v;     // adam v
  float* w32;   // fp32 master copy (mp variants)
  long n;
  long chunk0;  // first global chunk id of this tensor
};

constexpr int CHUNK = 65536;

DEVINL int find_tensor(const TensorDesc* descs, int ntens, long chunk) {
  // binary search over chunk0
  int lo = 0, hi = ntens - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (descs[mid].chunk0 <= chunk)
      lo = mid;
    else
      hi = mid - 1;
  }
  return lo;
}

// mode: 0 = sgd_mom (same-dtype), 1 = mp_sgd_mom (bf16 w + f32 master),
//       2 = adam, 3 = mp_adam
template <int MODE, bool WBF16>
__global__ __launch_bounds__(256) void multi_opt_kernel(
    const TensorDesc* __restrict__ descs, int ntens, long nchunks, float lr,
    float momentum, float beta1, float beta2, float eps, float wd,
    float rescale, float clip) {
  for (long c = blockIdx.x; c < nchunks; c += gridDim.x) {
    int ti = find_tensor(descs, ntens, c);
    const TensorDesc d = descs[ti];
    long off = (c - d.chunk0) * CHUNK;
    long end = min((long)CHUNK, d.n - off);
    for (long i = threadIdx.x; i < end; i += 256) {
      long k = off + i;
      float g;
      if (WBF16)
        g = bf2f(((const bf16*)d.g)[k]) * rescale;
      else
        g = ((const float*)d.g)[k] * rescale;
      if (clip > 0) g = fminf(fmaxf(g, -clip), clip);
      if (MODE == 0) {  // sgd_mom, weight dtype = storage dtype
        float w = WBF16 ? bf2f(((bf16*)d.w)[k]) : ((float*)d.w)[k];
        g += wd * w;
        float m = d.m[k] * momentum - lr * g;
        d.m[k] = m;
        w += m;
        if (WBF16)
          ((bf16*)d.w)[k] = f2bf(w);
        else
          ((float*)d.w)[k] = w;
      } else if (MODE == 1) {  // mp_sgd_mom
        float w = d.w32[k];
        g += wd * w;
        float m = d.m[k] * momentum - lr * g;
        d.m[k] = m;
        w += m;
        d.w32[k] = w;
        ((bf16*)d.w)[k] = f2bf(w);
      } else if (MODE == 2) {  // adam
        float w = WBF16 ? bf2f(((bf16*)d.w)[k]) : ((float*)d.w)[k];
        g += wd * w;
        float m = beta1 * d.m[k] + (1.f - beta1) * g;
        float v = beta2 * d.v[k] + (1.f - beta2) * g * g;
        d.m[k] = m;
        d.v[k] = v;
        w -= lr * m / (sqrtf(v) + eps);
        if (WBF16)
          ((bf16*)d.w)[k] = f2bf(w);
        else
          ((float*)d.w)[k] = w;
      } else {  // mp_adam
        float w = d.w32[k];
        g += wd * w;
        float m = beta1 * d.m[k] + (1.f - beta1) * g;
        float v = beta2 * d.v[k] + (1.f - beta2) * g * g;
        d.m[k] = m;
        d.v[k] = v;
        w -= lr * m / (sqrtf(v) + eps);
        d.w32[k] = w;
        ((bf16*)d.w)[k] = f2bf(w);
      }
    }
  }
}

// per-tensor sum of squares (LAMB r-norms / global-norm clipping)
template <bool BF16IN>
__global__ __launch_bounds__(256) void multi_sumsq_kernel(
    const TensorDesc* __restrict__ descs, int ntens, long nchunks,
    float* __restrict__ out) {
  __shared__ float scratch[4];
  for (long c = blockIdx.x; c < nchunks; c += gridDim.x) {
    int ti = find_tensor(descs, ntens, c);
    const TensorDesc d = descs[ti];
    long off = (c - d.chunk0) * CHUNK;
    long end = min((long)CHUNK, d.n - off);
    float ss = 0;
    for (long i = threadIdx.x; i < end; i += 256) {
      float v = BF16IN ? bf2f(((const bf16*)d.w)[off + i])
                       : ((const float*)d.w)[off + i];
      ss += v * v;
    }
#pragma unroll
    for (int o = 32; o > 0; o >>= 1) ss += __shfl_xor(ss, o, 64);
    int w = threadIdx.x / 64;
    if ((threadIdx.x & 63) == 0) scratch[w] = ss;
    __syncthreads();
    if (threadIdx.x == 0)
      atomicAdd(&out[ti],
                scratch[0] + scratch[1] + scratch[2] + scratch[3]);
    __syncthreads();
  }
}

void multi_opt(const TensorDesc* descs_dev, int ntens, long nchunks,
               int mode, bool wbf16, float lr, float momentum, float beta1,
               float beta2, float eps, float wd, float rescale, float clip,
               hipStream_t s) {
  long g = min(nchunks, (long)2048);
  dim3 grid((unsigned)(g < 1 ? 1 : g));
#define DISPATCH(M)                                                         \
  do {                                                                      \
    if (wbf16)                                                              \
     hipLaunchKernelGGL(( multi_opt_kernel<M, true>), dim3(grid), dim3(256), 0, s,                        \
          descs_dev, ntens, nchunks, lr, momentum, beta1, beta2, eps, wd,   \
          rescale, clip);                                                   \
    else                                                                    \
     hipLaunchKernelGGL(( multi_opt_kernel<M, false>), dim3(grid), dim3(256), 0, s,                       \
          descs_dev, ntens, nchunks, lr, momentum, beta1, beta2, eps, wd,   \
          rescale, clip);                                                   \
  } while (0)
  switch (mode) {
    case 0: DISPATCH(0); break;
    case 1: DISPATCH(1); break;
    case 2: DISPATCH(2); break;
    default: DISPATCH(3); break;
  }
#undef DISPATCH
}

void multi_sumsq(const TensorDesc* descs_dev, int ntens, long nchunks,
                 bool bf16in, float* out, hipStream_t s) {
  long g = min(nchunks, (long)2048);
  dim3 grid((unsigned)(g < 1 ? 1 : g));
  if (bf16in)
   hipLaunchKernelGGL(( multi_sumsq_kernel<true>), dim3(grid), dim3(256), 0, s, descs_dev, ntens, nchunks,
                                                  out);
  else
   hipLaunchKernelGGL(( multi_sumsq_kernel<false>), dim3(grid), dim3(256), 0, s, descs_dev, ntens,
                                                   nchunks, out);
}

}  // namespace mxa
