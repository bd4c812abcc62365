#include "hip/hip_runtime.h"
// NN-form MFMA GEMM for gfx950: A k-contiguous, B out-contiguous.
//
//   C[M][N] = sum_k A[M][k] * B[k][N]
//
// The third memory-layout form next to gemm_tn (both k-contiguous) and
// gemm_wg (both out-contiguous).  Having all three means NO transpose
// copies anywhere in a training step: every producer's natural layout is
// consumed directly (reference parity: linalg_gemm with transpose flags,
// which cuBLAS handles by strided reads — here each form is a dedicated
// CDNA4 kernel):
//   * Linear dgrad        : dx[m][in]  = dy[m][out] @ W[out][in]
//   * attention context   : O[tq][d]   = P[tq][tk] @ V[tk][d]
//   * attention dQ        : dQ[tq][d]  = dS[tq][tk] @ K[tk][d]
//   * generic _GemmTN da  : da[M][k]   = dy[M][N] @ B[N][k]
//
// A is staged into a [128 rows][64 k] XOR-swizzled image via
// global_load_lds and read with ds_read_b128 (exactly gemm_tn's A side).
// B is staged ROW-major into a [64 k][128 cols] image via global_load_lds
// and fragments come out of ds_read_b64_tr_b16 hardware-transpose reads
// (semantics probed in tools/probe_tr.hip; see gemm_wg.hip tr-variant).
//
// All three tensors carry a two-level batch map (outer*inner, used for
// per-(batch, head) attention slices of a [B, S, H*D] activation — the
// MI355X-native alternative to materializing [B,H,S,D] permute copies).
#include "common.h"

namespace mxa {

constexpr int NBM = 128, NBN = 128, NBK = 64, NTHREADS = 512;

// ---- A side (k-contiguous image, same scheme as gemm_tn) ----
DEVINL int nn_a_off(int row, int chunk16) {
  return row * NBK + ((chunk16 ^ (row & 7)) << 3);
}
DEVINL int nn_a_row(int p, int t) {
  // 512-thr blocks: 8 waves x 2 passes cover the 128 A rows
  return (((t >> 6) * 2 + p) << 3) + ((t & 63) >> 3);
}

// ---- B side (row-major [64][128] image, tr-read fragments) ----
DEVINL int nn_b_byte(int row, int cbyte) {
  return row * 256 + ((((cbyte >> 4) ^ ((row & 3) << 1)) << 4) | (cbyte & 15));
}

__global__ __launch_bounds__(NTHREADS, 2) void gemm_nn_core(
    const bf16* __restrict__ A, ZMap za, long lda, int M,
    const bf16* __restrict__ B, ZMap zb, long ldb, int N,
    bf16* __restrict__ Cout, ZMap zc, long ldc,
    int ksteps, const bf16* __restrict__ zpage) {
  __shared__ __align__(16) bf16 smem[2][2][NBM * NBK];

  const int ntn = (N + NBN - 1) / NBN;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = (bid / ntn) * NBM, bn = (bid % ntn) * NBN;
  const int t = threadIdx.x;
  const int z = blockIdx.z;
  const int lw = t / WAVE, ll = t % WAVE;

  // A row pointers (TN-style: pass p covers row0 + nn_a_row(p,t))
  const bf16* abase = A + za.off(z);
  const bf16* arp[2];
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int r = bm + nn_a_row(p, t);
    arp[p] = (r < M) ? abase + (long)r * lda : nullptr;
  }
  const int acg = (ll & 7) ^ (nn_a_row(0, t) & 7);  // row&7 == (ll>>3)&7, p-inv

  // B row pointers (tr-style: pass p covers k-row lw*8 + p*4 + (ll>>4))
  const bf16* bbase = B + zb.off(z);
  const int bcg = (ll & 15) ^ ((ll >> 4) << 1);
  int bcol = bn + bcg * 8;
  const bool bcolok = bcol + 8 <= N;
  const bf16* brp[2];
#pragma unroll
  for (int p = 0; p < 2; ++p)
    brp[p] = bbase + (long)(lw * 8 + p * 4 + (ll >> 4)) * ldb + bcol;

  auto stage_glds = [&](int buf, int ks) {
    long koff = (long)ks * NBK;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      unsigned lbase =
          __builtin_amdgcn_readfirstlane((unsigned)((lw * 2 + p) * 1024));
      auto ldsA = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smem[buf][0][0] + lbase);
      auto ldsB = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smem[buf][1][0] + lbase);
      auto gA = (const __attribute__((address_space(1))) unsigned int*)(
          arp[p] ? arp[p] + koff + acg * 8 : zpage + acg * 8);
      auto gB = (const __attribute__((address_space(1))) unsigned int*)(
          bcolok ? brp[p] + koff * ldb : zpage + bcg * 8);
      __builtin_amdgcn_global_load_lds(gA, ldsA, 16, 0, 0);
      __builtin_amdgcn_global_load_lds(gB, ldsB, 16, 0, 0);
    }
  };

  f32x4 acc[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // 8 waves as 4(M) x 2(N): each wave a 32x64 patch
  const int wm = (lw & 3) * 32, wn = (lw >> 2) * 64;
  const int lrow = ll & 15, lk = ll >> 4;
  const int ig = ll & 15, khi = ll >> 4;

  auto trread = [&](const bf16* img, int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)img + nn_b_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[2], bfr[4];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
        int row = wm + mf * 16 + lrow;
        af[mf] = *(const bf16x8*)&smem[buf][0][nn_a_off(row, chunk)];
      }
      int rbase = kk * 32 + khi * 8 + (ig >> 2);
      int cb = (ig & 3) * 8;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int cbyte = (wn + nf * 16) * 2 + cb;
        bf16x4 lo = trread(smem[buf][1], rbase, cbyte);
        bf16x4 hi = trread(smem[buf][1], rbase + 4, cbyte);
        bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
    }
  };

  stage_glds(0, 0);
  __syncthreads();
  for (int ks = 0; ks < ksteps; ++ks) {
    int buf = ks & 1;
    if (ks + 1 < ksteps) stage_glds(buf ^ 1, ks + 1);
    compute(buf);
    __syncthreads();
  }

  bf16* cz = Cout + zc.off(z);
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int gcol = bn + wn + nf * 16 + (ll & 15);
      if (gcol >= N) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + wm + mf * 16 + (ll >> 4) * 4 + i;
        if (grow >= M) continue;
        cz[(long)grow * ldc + gcol] = f2bf(acc[mf][nf][i]);
      }
    }
  }
}

// 64-wide-N variant: attention ctx/dQ have N = D = 64 — the 128-wide
// tile would run half-empty.  B image is [64 k][64 cols] (128-B rows),
// each wave computes a 64x32 quadrant (4 m-frags x 2 n-frags).
DEVINL int nn_b64_byte(int row, int cbyte) {
  return row * 128 + ((((cbyte >> 4) ^ ((row & 3) << 1)) << 4) | (cbyte & 15));
}

__global__ __launch_bounds__(NTHREADS, 2) void gemm_nn64_core(
    const bf16* __restrict__ A, ZMap za, long lda, int M,
    const bf16* __restrict__ B, ZMap zb, long ldb, int N,
    bf16* __restrict__ Cout, ZMap zc, long ldc,
    int ksteps, const bf16* __restrict__ zpage) {
  __shared__ __align__(16) bf16 smemA[2][NBM * NBK];
  __shared__ __align__(16) bf16 smemB[2][NBK * 64];

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = bid * NBM;
  const int t = threadIdx.x;
  const int z = blockIdx.z;
  const int lw = t / WAVE, ll = t % WAVE;

  const bf16* abase = A + za.off(z);
  const bf16* arp[2];
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int r = bm + nn_a_row(p, t);
    arp[p] = (r < M) ? abase + (long)r * lda : nullptr;
  }
  const int acg = (ll & 7) ^ (nn_a_row(0, t) & 7);

  // B staging: 8 waves, one 1-KiB issue each (8 rows of 128 B);
  // lane l -> row lw*8 + (l>>3), chunk l&7, source chunk XOR'd with
  // ((row&3)<<1).
  const bf16* bbase = B + zb.off(z);
  const int bcg = (ll & 7) ^ (((ll >> 3) & 3) << 1);
  int bcol = bcg * 8;
  const bool bcolok = bcol + 8 <= N;
  const bf16* brp0 = bbase + (long)(lw * 8 + (ll >> 3)) * ldb + bcol;

  auto stage_glds = [&](int buf, int ks) {
    long koff = (long)ks * NBK;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      unsigned lbase =
          __builtin_amdgcn_readfirstlane((unsigned)((lw * 2 + p) * 1024));
      auto ldsA = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smemA[buf][0] + lbase);
      auto gA = (const __attribute__((address_space(1))) unsigned int*)(
          arp[p] ? arp[p] + koff + acg * 8 : zpage + acg * 8);
      __builtin_amdgcn_global_load_lds(gA, ldsA, 16, 0, 0);
    }
    {
      unsigned lbase =
          __builtin_amdgcn_readfirstlane((unsigned)(lw * 1024));
      auto ldsB = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smemB[buf][0] + lbase);
      auto gB = (const __attribute__((address_space(1))) unsigned int*)(
          bcolok ? brp0 + koff * ldb : zpage + bcg * 8);
      __builtin_amdgcn_global_load_lds(gB, ldsB, 16, 0, 0);
    }
  };

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // 8 waves as 4(M) x 2(N): each wave a 32x32 patch
  const int wm = (lw & 3) * 32, wn = (lw >> 2) * 32;
  const int lrow = ll & 15, lk = ll >> 4;
  const int ig = ll & 15, khi = ll >> 4;

  auto trread = [&](const bf16* img, int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)img +
        nn_b64_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[2], bfr[2];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
        int row = wm + mf * 16 + lrow;
        af[mf] = *(const bf16x8*)&smemA[buf][nn_a_off(row, chunk)];
      }
      int rbase = kk * 32 + khi * 8 + (ig >> 2);
      int cb = (ig & 3) * 8;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int cbyte = (wn + nf * 16) * 2 + cb;
        bf16x4 lo = trread(smemB[buf], rbase, cbyte);
        bf16x4 hi = trread(smemB[buf], rbase + 4, cbyte);
        bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
    }
  };

  stage_glds(0, 0);
  __syncthreads();
  for (int ks = 0; ks < ksteps; ++ks) {
    int buf = ks & 1;
    if (ks + 1 < ksteps) stage_glds(buf ^ 1, ks + 1);
    compute(buf);
    __syncthreads();
  }

  bf16* cz = Cout + zc.off(z);
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      int gcol = wn + nf * 16 + (ll & 15);
      if (gcol >= N) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + wm + mf * 16 + (ll >> 4) * 4 + i;
        if (grow >= M) continue;
        cz[(long)grow * ldc + gcol] = f2bf(acc[mf][nf][i]);
      }
    }
  }
}

void gemm_nn_bf16(const bf16* A, ZMap za, long lda, int M,
                  const bf16* B, ZMap zb, long ldb, int N,
                  bf16* C, ZMap zc, long ldc, int K, int nz,
                  const bf16* zpage, hipStream_t stream) {
  int ksteps = K / NBK;
  if (N <= 64) {
    dim3 grid((unsigned)cdiv_h(M, NBM), 1, nz);
   hipLaunchKernelGGL(( gemm_nn64_core), dim3(grid), dim3(NTHREADS), 0, stream, A, za, lda, M, B, zb,
                                                  ldb, N, C, zc, ldc,
                                                  ksteps, zpage);
    return;
  }
  long bxy = (long)cdiv_h(M, NBM) * cdiv_h(N, NBN);
  dim3 grid((unsigned)bxy, 1, nz);
 hipLaunchKernelGGL(( gemm_nn_core), dim3(grid), dim3(NTHREADS), 0, stream, A, za, lda, M, B, zb, ldb, N,
                                              C, zc, ldc, ksteps, zpage);
}

}  // namespace mxa
