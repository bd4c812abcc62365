// Generic elementwise / broadcast / reduce launcher for gfx950 — the
// framework-owned long-tail op surface (reference parity:
// src/operator/mxnet_op.h Kernel<OP, gpu>::Launch + mshadow_op.h functor
// zoo + broadcast_reduce-inl.cuh).  One grid-stride template serves every
// functor; bf16 paths use bf16x8 vector IO on contiguous tensors and a
// strided multi-index fallback for broadcasts (up to 6 dims).
//
// fp32 math inside, bf16 storage (same contract as the rest of the
// kernel library); gradients are separate functor launches wired from
// python autograd.Functions.
#include "common.h"

namespace mxa {

// ---------------------------------------------------------------------------
// functor tables (enum shared with bindings.hip / ops python)
// ---------------------------------------------------------------------------

enum UnaryOp {
  U_RELU = 0, U_SIGMOID, U_TANH, U_EXP, U_LOG, U_SQRT, U_RSQRT, U_SQUARE,
  U_ABS, U_NEG, U_RECIP, U_SIN, U_COS, U_ERF, U_LOG1P, U_EXPM1, U_FLOOR,
  U_CEIL, U_ROUND, U_SIGN, U_SOFTSIGN, U_COUNT
};

DEVINL float unary_apply(int op, float x) {
  switch (op) {
    case U_RELU: return x > 0.f ? x : 0.f;
    case U_SIGMOID: return 1.f / (1.f + __expf(-x));
    case U_TANH: return tanhf(x);
    case U_EXP: return __expf(x);
    case U_LOG: return __logf(x);
    case U_SQRT: return sqrtf(x);
    case U_RSQRT: return rsqrtf(x);
    case U_SQUARE: return x * x;
    case U_ABS: return fabsf(x);
    case U_NEG: return -x;
    case U_RECIP: return 1.f / x;
    case U_SIN: return __sinf(x);
    case U_COS: return __cosf(x);
    case U_ERF: return erff(x);
    case U_LOG1P: return log1pf(x);
    case U_EXPM1: return expm1f(x);
    case U_FLOOR: return floorf(x);
    case U_CEIL: return ceilf(x);
    case U_ROUND: return rintf(x);
    case U_SIGN: return x > 0.f ? 1.f : (x < 0.f ? -1.f : 0.f);
    case U_SOFTSIGN: return x / (1.f + fabsf(x));
  }
  return x;
}

// dX as a function of (x, y, dy) — the mshadow_op *_grad functors
DEVINL float unary_grad(int op, float x, float y, float dy) {
  switch (op) {
    case U_RELU: return x > 0.f ? dy : 0.f;
    case U_SIGMOID: return dy * y * (1.f - y);
    case U_TANH: return dy * (1.f - y * y);
    case U_EXP: return dy * y;
    case U_LOG: return dy / x;
    case U_SQRT: return dy * 0.5f / y;
    case U_RSQRT: return dy * -0.5f * y / x;
    case U_SQUARE: return dy * 2.f * x;
    case U_ABS: return x >= 0.f ? dy : -dy;
    case U_NEG: return -dy;
    case U_RECIP: return -dy * y * y;
    case U_SIN: return dy * __cosf(x);
    case U_COS: return -dy * __sinf(x);
    case U_ERF: return dy * 1.1283791671f * __expf(-x * x);
    case U_LOG1P: return dy / (1.f + x);
    case U_EXPM1: return dy * __expf(x);
    case U_SOFTSIGN: {
      float d = 1.f + fabsf(x);
      return dy / (d * d);
    }
  }
  return 0.f;  // floor/ceil/round/sign: zero gradient
}

enum BinaryOp {
  B_ADD = 0, B_SUB, B_MUL, B_DIV, B_POW, B_MAX, B_MIN, B_COUNT
};

DEVINL float binary_apply(int op, float a, float b) {
  switch (op) {
    case B_ADD: return a + b;
    case B_SUB: return a - b;
    case B_MUL: return a * b;
    case B_DIV: return a / b;
    case B_POW: return powf(a, b);
    case B_MAX: return fmaxf(a, b);
    case B_MIN: return fminf(a, b);
  }
  return 0.f;
}

// ---------------------------------------------------------------------------
// contiguous kernels (bf16x8 vector IO; tail handled scalar)
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void ew_unary_kernel(
    int op, const bf16* __restrict__ x, bf16* __restrict__ y, long n8,
    long ntail) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n8; i += (long)gridDim.x * blockDim.x) {
    bf16x8 v = *(const bf16x8*)&x[i * 8];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(unary_apply(op, bf2f(v[j])));
    *(bf16x8*)&y[i * 8] = o;
  }
  // tail (single block 0 re-walks the stragglers)
  if (blockIdx.x == 0) {
    for (long k = n8 * 8 + threadIdx.x; k < n8 * 8 + ntail;
         k += blockDim.x)
      y[k] = f2bf(unary_apply(op, bf2f(x[k])));
  }
}

__global__ __launch_bounds__(256) void ew_unary_grad_kernel(
    int op, const bf16* __restrict__ x, const bf16* __restrict__ y,
    const bf16* __restrict__ dy, bf16* __restrict__ dx, long n8,
    long ntail) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n8; i += (long)gridDim.x * blockDim.x) {
    bf16x8 vx = *(const bf16x8*)&x[i * 8];
    bf16x8 vy = *(const bf16x8*)&y[i * 8];
    bf16x8 vd = *(const bf16x8*)&dy[i * 8];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bf(unary_grad(op, bf2f(vx[j]), bf2f(vy[j]), bf2f(vd[j])));
    *(bf16x8*)&dx[i * 8] = o;
  }
  if (blockIdx.x == 0) {
    for (long k = n8 * 8 + threadIdx.x; k < n8 * 8 + ntail;
         k += blockDim.x)
      dx[k] = f2bf(
          unary_grad(op, bf2f(x[k]), bf2f(y[k]), bf2f(dy[k])));
  }
}

// broadcast binary: strided multi-index walk (up to 6 dims).  Shapes are
// pre-aligned on the host (size-1 dims get stride 0) — the reference's
// broadcast_kernel computed the same per-element unravel.
struct BShape {
  int ndim;
  long shape[6];
  long sa[6], sb[6];
};

__global__ __launch_bounds__(256) void ew_binary_bcast_kernel(
    int op, const bf16* __restrict__ a, const bf16* __restrict__ b,
    bf16* __restrict__ y, BShape s, long n) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    long rem = i, oa = 0, ob = 0;
#pragma unroll
    for (int d = 5; d >= 0; --d) {
      if (d >= s.ndim) continue;
      long idx = rem % s.shape[d];
      rem /= s.shape[d];
      oa += idx * s.sa[d];
      ob += idx * s.sb[d];
    }
    y[i] = f2bf(binary_apply(op, bf2f(a[oa]), bf2f(b[ob])));
  }
}

// contiguous same-shape binary (fast path)
__global__ __launch_bounds__(256) void ew_binary_kernel(
    int op, const bf16* __restrict__ a, const bf16* __restrict__ b,
    bf16* __restrict__ y, long n8, long ntail) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n8; i += (long)gridDim.x * blockDim.x) {
    bf16x8 va = *(const bf16x8*)&a[i * 8];
    bf16x8 vb = *(const bf16x8*)&b[i * 8];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bf(binary_apply(op, bf2f(va[j]), bf2f(vb[j])));
    *(bf16x8*)&y[i * 8] = o;
  }
  if (blockIdx.x == 0) {
    for (long k = n8 * 8 + threadIdx.x; k < n8 * 8 + ntail;
         k += blockDim.x)
      y[k] = f2bf(binary_apply(op, bf2f(a[k]), bf2f(b[k])));
  }
}

// ---------------------------------------------------------------------------
// reductions over the LAST axis (rows x D) — wave-per-row, fp32 accum
// (reference MXNET_SAFE_ACCUMULATION); 0 = sum, 1 = max, 2 = min
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void ew_reduce_last_kernel(
    int op, const bf16* __restrict__ x, float* __restrict__ out, long rows,
    long D) {
  const int w = threadIdx.x >> 6, l = threadIdx.x & 63;
  for (long r = (long)blockIdx.x * 4 + w; r < rows;
       r += 4L * (long)gridDim.x) {
    const bf16* xp = x + r * D;
    float acc = op == 0 ? 0.f : (op == 1 ? -3.0e38f : 3.0e38f);
    for (long k = l * 8; k + 8 <= D; k += 64 * 8) {
      bf16x8 v = *(const bf16x8*)&xp[k];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v[j]);
        acc = op == 0 ? acc + f : (op == 1 ? fmaxf(acc, f) : fminf(acc, f));
      }
    }
    // scalar tail
    long tail0 = (D / (64 * 8)) * 64 * 8 + l * 8;
    (void)tail0;
    for (long k = (D / 8) * 8 + l; k < D; k += 64) {
      float f = bf2f(xp[k]);
      acc = op == 0 ? acc + f : (op == 1 ? fmaxf(acc, f) : fminf(acc, f));
    }
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      float o = __shfl_xor(acc, off, 64);
      acc = op == 0 ? acc + o : (op == 1 ? fmaxf(acc, o) : fminf(acc, o));
    }
    if (l == 0) out[r] = acc;
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static int ew_blocks(long items) {
  long b = (items + 255) / 256;
  return (int)(b > 4096 ? 4096 : (b < 1 ? 1 : b));
}

void ew_unary(int op, const bf16* x, bf16* y, long n, hipStream_t s) {
  long n8 = n / 8;
  ew_unary_kernel<<<ew_blocks(n8 ? n8 : 1), 256, 0, s>>>(op, x, y, n8,
                                                         n - n8 * 8);
}

void ew_unary_grad(int op, const bf16* x, const bf16* y, const bf16* dy,
                   bf16* dx, long n, hipStream_t s) {
  long n8 = n / 8;
  ew_unary_grad_kernel<<<ew_blocks(n8 ? n8 : 1), 256, 0, s>>>(
      op, x, y, dy, dx, n8, n - n8 * 8);
}

void ew_binary(int op, const bf16* a, const bf16* b, bf16* y, long n,
               hipStream_t s) {
  long n8 = n / 8;
  ew_binary_kernel<<<ew_blocks(n8 ? n8 : 1), 256, 0, s>>>(op, a, b, y, n8,
                                                          n - n8 * 8);
}

void ew_binary_bcast(int op, const bf16* a, const bf16* b, bf16* y,
                     int ndim, const long* shape, const long* sa,
                     const long* sb, long n, hipStream_t s) {
  BShape bs;
  bs.ndim = ndim;
  for (int d = 0; d < 6; ++d) {
    bs.shape[d] = d < ndim ? shape[d] : 1;
    bs.sa[d] = d < ndim ? sa[d] : 0;
    bs.sb[d] = d < ndim ? sb[d] : 0;
  }
  ew_binary_bcast_kernel<<<ew_blocks(n), 256, 0, s>>>(op, a, b, y, bs, n);
}

void ew_reduce_last(int op, const bf16* x, float* out, long rows, long D,
                    hipStream_t s) {
  long b = (rows + 3) / 4;
  ew_reduce_last_kernel<<<(unsigned)(b > 4096 ? 4096 : b), 256, 0, s>>>(
      op, x, out, rows, D);
}

}  // namespace mxa
