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

// ---------------------------------------------------------------------------
// 256^2 window variant (the 2-barrier-per-K-tile discipline measured
// +23% on gemm_tn8g_core): fixed 128x64 wave tiles, A fragments via
// plain b128 reads of the row-XOR image, B fragments via tr reads of
// row-major [64 k][128 n] images, all register-resident before a
// barrier-free 64-MFMA setprio section.  Same issue/vmcnt accounting as
// tn8g: B halves of tile+1 in the pre-zone (other buffer), A halves of
// tile+2 inside the MFMA section (this buffer, reads published by the
// opening lgkmcnt(0)+barrier), one vmcnt(4) per window.
// ---------------------------------------------------------------------------

// tr-image swizzle with a WIDER key than nn_b_byte: the fragment read
// set spans rows {q + 4h + 8*khi} — row&3 alone leaves the four khi
// groups on identical chunk columns (PMC: 100M 4-way conflicts at
// 8192^3); folding (row>>3)&3 into the key separates them.
DEVINL int nn_b2_byte(int row, int cbyte) {
  int key = ((row & 3) << 1) ^ ((row >> 3) & 3);
  return row * 256 + ((((cbyte >> 4) ^ key) << 4) | (cbyte & 15));
}

__global__ __launch_bounds__(NTHREADS, 1) void gemm_nn8g_core(
    const bf16* __restrict__ A, ZMap za, long lda, int M,
    const bf16* __restrict__ B, ZMap zb, long ldb, int N,
    bf16* __restrict__ Cout, ZMap zc, long ldc,
    int ksteps, const bf16* __restrict__ zpage) {
  __shared__ __align__(16) bf16 smemA[2][2][128 * 64];
  __shared__ __align__(16) bf16 smemB[2][2][64 * 128];

  const int ntn = (N + 255) / 256;
  const int ntm = (M + 255) / 256;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm, tnn;
  if ((ntm & 3) == 0 && ntm * ntn > 256) {  // 2D XCD patches (tn8g)
    tm = (bid / (4 * ntn)) * 4 + (bid & 3);
    tnn = (bid >> 2) % ntn;
  } else {
    tm = bid / ntn;
    tnn = bid % ntn;
  }
  const int bm = tm * 256, bn = tnn * 256;
  const int t = threadIdx.x;
  const int z = blockIdx.z;
  const int lw = t / WAVE, ll = t % WAVE;

  // A loader: 4 row-slots (half = p>>1, glds pass j = p&1), tn8 scheme
  const bf16* abase = A + za.off(z);
  const bf16* arp[4];
  int acg[4];
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    int local = (p >> 1) * 128 + lw * 16 + (p & 1) * 8 + (ll >> 3);
    int r = bm + local;
    arp[p] = (r < M) ? abase + (long)r * lda : nullptr;
    acg[p] = (ll & 7) ^ (local & 7);
  }
  auto issue_a = [&](int half, int tile) {
    long koff = (long)tile * NBK;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int p = half * 2 + j;
      unsigned seg = __builtin_amdgcn_readfirstlane(
          (unsigned)(lw * 2048 + j * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smemA[tile & 1][half][0] +
          seg);
      const bf16* gp = arp[p] ? arp[p] + koff + acg[p] * 8
                              : zpage + acg[p] * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
          0, 0);
    }
  };

  // B loader: [64 k][128 n] halves, 256-B rows, tr key (row&3)<<1.
  // Stage pass p covers rows p*32 + (lw*4 + ll>>4); per-lane source
  // column clamped against N (zpage holds 128 zero bf16).
  const int bsrow = lw * 4 + (ll >> 4);
  // per-pass source chunk for nn_b2_byte's wider key (row bits 0-1 and
  // 3-4; pass p only sets row bit 5, so the key is in fact p-invariant
  // -- kept per-pass for clarity, the unroller folds the duplicates)
  const bf16* bbase = B + zb.off(z);
  int bcg[2];
  const bf16* brp0[2];
  bool bok[2][2];
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int row = p * 32 + bsrow;
    int key = ((row & 3) << 1) ^ ((row >> 3) & 3);
    bcg[p] = (ll & 15) ^ key;
    brp0[p] = bbase + (long)row * ldb + bn + bcg[p] * 8;
#pragma unroll
    for (int h = 0; h < 2; ++h)
      bok[p][h] = bn + h * 128 + bcg[p] * 8 + 8 <= N;
  }
  auto issue_b = [&](int half, int tile) {
    long koff = (long)tile * NBK * ldb + (long)half * 128;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      unsigned seg = __builtin_amdgcn_readfirstlane(
          (unsigned)((p * 32 + lw * 4) * 256));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smemB[tile & 1][half][0] +
          seg);
      const bf16* gp = bok[p][half] ? brp0[p] + koff : zpage + bcg[p] * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
          0, 0);
    }
  };

  const int lrow = ll & 15, lk = ll >> 4;
  const int ig = ll & 15, khi = ll >> 4;
  const int wr = lw >> 2;          // A half (M)
  const int wc = lw & 3;           // N quarter
  const int bh = wc >> 1;          // B half
  const int bno = (wc & 1) * 64;   // N offset within the half

  f32x4 acc[8][4];
#pragma unroll
  for (int mf = 0; mf < 8; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) acc[mf][nf] = f32x4{0.f, 0.f, 0.f, 0.f};

  bf16x8 a[8][2], b[4][2];

  auto trread = [&](const bf16* img, int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)img +
        nn_b2_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };

  // prologue: tile0 A+B + tile1 A (vmcnt(4) leaves tile1's A in flight)
  issue_a(0, 0);
  issue_a(1, 0);
  issue_b(0, 0);
  issue_b(1, 0);
  if (1 < ksteps) {
    issue_a(0, 1);
    issue_a(1, 1);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int tile = 0; tile < ksteps; ++tile) {
    const bf16* As = smemA[tile & 1][wr];
    const bf16* Bs = smemB[tile & 1][bh];
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        a[mf][kk] = *(const bf16x8*)&As[nn_a_off(mf * 16 + lrow,
                                                 kk * 4 + lk)];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int rbase = kk * 32 + khi * 8 + (ig >> 2);
      int cb = (ig & 3) * 8;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int cbyte = (bno + nf * 16) * 2 + cb;
        bf16x4 lo = trread(Bs, rbase, cbyte);
        bf16x4 hi = trread(Bs, rbase + 4, cbyte);
        b[nf][kk] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
    }
    if (tile + 1 < ksteps) {
      issue_b(0, tile + 1);
      issue_b(1, tile + 1);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      if (kk == 0 && tile + 2 < ksteps) issue_a(0, tile + 2);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mf][kk], b[nf][kk], acc[mf][nf], 0, 0, 0);
      if (kk == 0 && tile + 2 < ksteps) issue_a(1, tile + 2);
#pragma unroll
      for (int mf = 4; mf < 8; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mf][kk], b[nf][kk], acc[mf][nf], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    if (tile + 2 < ksteps)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (tile + 1 < ksteps)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  bf16* cz = Cout + zc.off(z);
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int gcol = bn + wc * 64 + nf * 16 + (ll & 15);
      if (gcol >= N) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + wr * 128 + mf * 16 + (ll >> 4) * 4 + i;
        if (grow >= M) continue;
        cz[(long)grow * ldc + gcol] = f2bf(acc[mf][nf][i]);
      }
    }
  }
}

static long nn8_minblocks() {
  static const long v = [] {
    const char* e = getenv("MXA_NN8_MINBLOCKS");
    // 160: BERT's 192-tile K=3072 dx shapes win on the window kernel
    // (506 -> 517k tok/s e2e); tbig unchanged
    return e ? atol(e) : 160L;
  }();
  return v;
}

static bool nn8_useg() {
  // default ON for K >= 1024 grids of >= 240 tiles (A/B: +7-30% — 1041
  // vs 799 TF at 8192^3, +12% at 16384x4096x1024, +7.5% at the ResNet
  // dgrad shape 200704x512x2048; K=768 shapes LOSE ~15% and stay on
  // gemm_nn_core).  MXA_NN8G=0 reverts.
  static const bool v = [] {
    const char* e = getenv("MXA_NN8G");
    return !e || e[0] != '0';
  }();
  return v;
}

void gemm_nn_bf16(const bf16* A, ZMap za, long lda, int M,
                  const bf16* B, ZMap zb, long ldb, int N,
                  bf16* C, ZMap zc, long ldc, int K, int nz,
                  const bf16* zpage, hipStream_t stream) {
  int ksteps = K / NBK;
  if (N <= 64) {
    dim3 grid((unsigned)cdiv_h(M, NBM), 1, nz);
    gemm_nn64_core<<<grid, NTHREADS, 0, stream>>>(A, za, lda, M, B, zb,
                                                  ldb, N, C, zc, ldc,
                                                  ksteps, zpage);
    return;
  }
  long bxy256 = (long)cdiv_h(M, 256) * cdiv_h(N, 256) * (long)nz;
  if (nn8_useg() && M >= 256 && N >= 256 && bxy256 >= nn8_minblocks() &&
      K >= 1024) {
    dim3 grid((unsigned)(cdiv_h(M, 256) * cdiv_h(N, 256)), 1, nz);
    gemm_nn8g_core<<<grid, NTHREADS, 0, stream>>>(A, za, lda, M, B, zb,
                                                  ldb, N, C, zc, ldc,
                                                  ksteps, zpage);
    return;
  }
  long bxy = (long)cdiv_h(M, NBM) * cdiv_h(N, NBN);
  dim3 grid((unsigned)bxy, 1, nz);
  gemm_nn_core<<<grid, NTHREADS, 0, stream>>>(A, za, lda, M, B, zb, ldb, N,
                                              C, zc, ldc, ksteps, zpage);
}

}  // namespace mxa
