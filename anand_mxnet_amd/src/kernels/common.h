// Common device helpers for the CDNA4 (gfx950) kernel library.
// MI355X-native: wave64, 4 SIMD-32/CU, MFMA bf16 32x32x16 / 16x16x32,
// LDS 160 KiB/CU. See /opt/skills/guides/cdna_hip_programming.md for the
// idioms these kernels follow (T1 XCD swizzle, T2 LDS XOR swizzle, T14
// async-stage split, G13 vectorized bf16 loads).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
// slot count for the fused BN-stats atomic accumulators ([SLOTS][2][C])
#define STATS_SLOTS 32
#define DEVINL __device__ __forceinline__

typedef __bf16 bf16;
typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x2 __attribute__((ext_vector_type(2)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef int i32x4 __attribute__((ext_vector_type(4)));
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef unsigned int u32;
typedef unsigned long long u64;

DEVINL float bf2f(bf16 x) { return __bfloat162float(x); }
DEVINL bf16 f2bf(float x) { return __float2bfloat16(x); }

// Two-level batch offset map: z = zo * zdiv + zi -> zo*souter + zi*sinner.
// Used for per-(batch, head) slices of a [B, S, H*D] activation so batched
// attention GEMMs read heads IN PLACE (no [B,H,S,D] permute copies).
// ZMap{1, stride, 0} is a plain contiguous batch; ZMap{1, 0, 0} unbatched.
struct ZMap {
  int zdiv;
  long souter, sinner;
  DEVINL long off(int z) const {
    return (long)(z / zdiv) * souter + (long)(z % zdiv) * sinner;
  }
};

// splitmix64-finalizer counter hash (dropout RNG; statistically strong,
// stateless — reference parity: philox counter RNG in dropout.cu)
DEVINL u32 hash_u32(u64 seed, u64 idx) {
  u64 z = seed + 0x9e3779b97f4a7c15ULL * (idx + 1);
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ULL;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebULL;
  return (u32)(z >> 32);
}

DEVINL u64 seed_finalize(u64 z) {
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ULL;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebULL;
  return z ^ (z >> 31);
}

// ceil-div
DEVINL constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }
static inline int cdiv_h(int a, int b) { return (a + b - 1) / b; }

// Bijective XCD-aware block remap (guide §5: XCD swizzle must be bijective).
// Consecutive output ids land on one XCD so neighbouring tiles share L2.
DEVINL int xcd_swizzle(int bid, int nblocks) {
  const int NXCD = 8;
  if (nblocks < 2 * NXCD) return bid;
  int q = nblocks / NXCD, r = nblocks % NXCD;
  int xcd = bid % NXCD, pos = bid / NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
}

// Grid-stride elementwise launch geometry (guide G11): 256-thr blocks,
// cap blocks, stride the rest.
static inline dim3 ew_grid(long long n_vec, int block = 256,
                           int max_blocks = 2048) {
  long long b = (n_vec + block - 1) / block;
  if (b > max_blocks) b = max_blocks;
  if (b < 1) b = 1;
  return dim3((unsigned)b);
}

#define HIP_CHECK(cmd)                                                       \
  do {                                                                       \
    hipError_t e_ = (cmd);                                                   \
    if (e_ != hipSuccess) {                                                  \
      throw std::runtime_error(std::string("HIP error: ") +                  \
                               hipGetErrorString(e_) + " at " __FILE__ ":" + \
                               std::to_string(__LINE__));                    \
    }                                                                        \
  } while (0)
