// 256x256-tile 8-phase MFMA GEMM for gfx950 — the deep-pipelined dense
// path (guide §5 "256² 8-phase template"): 512 threads (8 waves as 2Mx4N),
// BK=64, half-tile glds ring (2 tile-buffers x 2 halves per operand,
// 128 KiB LDS), COUNTED s_waitcnt vmcnt across raw barriers (never 0 in
// the main loop), quadrant-phased compute with s_setprio around each MFMA
// cluster, T2 source-side XOR swizzle, T1 bijective XCD remap.
//
// Phase schedule (tile T, quadrants q = (A-half, B-half)):
//   p0 (A0,B0): issue T+1.A0 ; vmcnt(6) ; barrier ; 16 MFMA
//   p1 (A1,B0): issue T+1.B0 ; vmcnt(6) ; barrier ; 16 MFMA
//   p2 (A1,B1): issue T+1.A1 ; vmcnt(6) ; barrier ; 16 MFMA
//   p3 (A0,B1): issue T+1.B1 ;  (no wait) ; barrier ; 16 MFMA
// so three halves stay in flight at every wait (vmcnt counts 2 glds per
// half).  Slot of tile T's half is reused by tile T+2, whose issue happens
// after T's last-reader barrier.
//
// Used for the dense NLP GEMMs (Linear/FFN/attention, K % 128 == 0);
// convolution keeps the 128² kernel in gemm_tn.hip.
#include "common.h"

namespace mxa {

constexpr int T8_BM = 256, T8_BN = 256, T8_BK = 64, T8_THREADS = 512;

DEVINL int t8_lds_byte(int row, int chunk) {
  // byte offset within a [128][64] bf16 half-image, T2 row-XOR swizzle
  return row * 128 + ((chunk ^ (row & 7)) << 4);
}

struct Plain8Loader {
  const bf16* base;
  const bf16* zpage;
  int rows;
  long ld;
  long stride_z;
  // device state: 4 row-slots (half a = p>>1, instr j = p&1)
  const bf16* rp[4];
  int cg[4];

  DEVINL void init(int row0, int t, int z) {
    const bf16* b = base + (long)z * stride_z;
    int w = t / WAVE, l = t % WAVE;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      int local = (p >> 1) * 128 + w * 16 + (p & 1) * 8 + (l >> 3);
      int r = row0 + local;
      rp[p] = (r < rows) ? b + (long)r * ld : nullptr;
      cg[p] = (l & 7) ^ (local & 7);  // source chunk for lane's slot
    }
  }
  DEVINL const bf16* src(int p, long koff) const {
    return rp[p] ? rp[p] + koff + cg[p] * 8 : zpage + cg[p] * 8;
  }
};

template <bool RELU, bool BIAS>
__global__ __launch_bounds__(T8_THREADS, 1) void gemm_tn8_core(
    Plain8Loader aload, Plain8Loader bload, bf16* __restrict__ Cout,
    const float* __restrict__ bias, int M, int N, long ldc, long strideC,
    int ksteps, long kstride) {
  // [tilebuf][op][half][128x64 bf16]
  __shared__ __align__(16) bf16 smem[2][2][2][128 * 64];

  const int ntn = (N + T8_BN - 1) / T8_BN;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = (bid / ntn) * T8_BM, bn = (bid % ntn) * T8_BN;
  const int t = threadIdx.x;
  const int z = blockIdx.z;
  const int w = t / WAVE, l = t % WAVE;

  aload.init(bm, t, z);
  bload.init(bn, t, z);

  // one glds pair: half `hf` (0=A0,1=B0,2=A1,3=B1 order by issue index) of
  // tile `tile` into its ring slot.
  auto issue_half = [&](int issue_idx, int tile) {
    int op = issue_idx & 1;          // A0,B0,A1,B1
    int half = issue_idx >> 1;
    long koff = (long)tile * kstride;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int p = half * 2 + j;  // loader slot
      unsigned seg = __builtin_amdgcn_readfirstlane(
          (unsigned)(w * 2048 + j * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smem[tile & 1][op][half]
              [0] + seg);
      const bf16* gp = op ? bload.src(p, koff) : aload.src(p, koff);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
          0, 0);
    }
  };

  f32x4 acc[4][8];  // [quadrant][mf*2+nf]
#pragma unroll
  for (int q = 0; q < 4; ++q)
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[q][i] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int lrow = l & 15, lk = l >> 4;
  const int wmq = (w >> 2) * 64;  // within-quadrant M offset of this wave
  const int wnq = (w & 3) * 32;   // within-quadrant N offset

  // one phase: quadrant (qa, qb) of tile `tile`
  auto phase = [&](int tile, int qa, int qb, int q, int wait6) {
    if (wait6 == 1)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if (wait6 == 0)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    const bf16* As = smem[tile & 1][0][qa];
    const bf16* Bs = smem[tile & 1][1][qb];
    bf16x8 af[4][2], bfr[2][2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
        af[mf][kk] = *(const bf16x8*)((const char*)As +
                                      t8_lds_byte(wmq + mf * 16 + lrow,
                                                  chunk));
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        bfr[nf][kk] = *(const bf16x8*)((const char*)Bs +
                                       t8_lds_byte(wnq + nf * 16 + lrow,
                                                   chunk));
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[q][mf * 2 + nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf][kk], bfr[nf][kk], acc[q][mf * 2 + nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  };

  // prologue: tile 0's 4 halves
#pragma unroll
  for (int h = 0; h < 4; ++h) issue_half(h, 0);

  // quadrant order: (A0,B0) (A1,B0) (A1,B1) (A0,B1); q index = phase index
  for (int tile = 0; tile < ksteps; ++tile) {
    int nxt = tile + 1;
    bool pre = nxt < ksteps;
    // p0
    if (pre) issue_half(0, nxt);
    phase(tile, 0, 0, 0, pre ? 1 : 0);
    // p1
    if (pre) issue_half(1, nxt);
    phase(tile, 1, 0, 1, pre ? 1 : -1);
    // p2
    if (pre) issue_half(2, nxt);
    phase(tile, 1, 1, 2, pre ? 1 : -1);
    // p3
    if (pre) issue_half(3, nxt);
    phase(tile, 0, 1, 3, -1);
  }

  // epilogue
  bf16* cz = Cout + (long)z * strideC;
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    int qa = (q == 1 || q == 2) ? 1 : 0;  // A-half of quadrant q
    int qb = (q >= 2) ? 1 : 0;            // B-half
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int gcol = bn + qb * 128 + wnq + nf * 16 + (l & 15);
        if (gcol >= N) continue;
        float bv = BIAS ? bias[gcol] : 0.f;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int grow = bm + qa * 128 + wmq + mf * 16 + (l >> 4) * 4 + i;
          if (grow >= M) continue;
          float v = acc[q][mf * 2 + nf][i] + bv;
          if (RELU) v = v > 0.f ? v : 0.f;
          cz[(long)grow * ldc + gcol] = f2bf(v);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 32x32x16 variant: same tile/ring/phase structure, but each phase is
// 8 x v_mfma_f32_32x32x16_bf16 instead of 16 x 16x16x32 — half the MFMA
// issue slots (PMC showed 48% SQ_WAIT_INST_ANY on the 16x16 kernel) and
// the 32x32 shape's µbench ceiling is ~15% higher (2382 vs 2075 TF).
// C/D layout: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5).
// ---------------------------------------------------------------------------

template <bool RELU, bool BIAS>
__global__ __launch_bounds__(T8_THREADS, 1) void gemm_tn8x32_core(
    Plain8Loader aload, Plain8Loader bload, bf16* __restrict__ Cout,
    const float* __restrict__ bias, int M, int N, long ldc, long strideC,
    int ksteps, long kstride) {
  __shared__ __align__(16) bf16 smem[2][2][2][128 * 64];

  const int ntn = (N + T8_BN - 1) / T8_BN;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = (bid / ntn) * T8_BM, bn = (bid % ntn) * T8_BN;
  const int t = threadIdx.x;
  const int z = blockIdx.z;
  const int w = t / WAVE, l = t % WAVE;

  aload.init(bm, t, z);
  bload.init(bn, t, z);

  auto issue_half = [&](int issue_idx, int tile) {
    int op = issue_idx & 1;
    int half = issue_idx >> 1;
    long koff = (long)tile * kstride;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int p = half * 2 + j;
      unsigned seg = __builtin_amdgcn_readfirstlane(
          (unsigned)(w * 2048 + j * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smem[tile & 1][op][half]
              [0] + seg);
      const bf16* gp = op ? bload.src(p, koff) : aload.src(p, koff);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
          0, 0);
    }
  };

  f32x16 acc[4][2];  // [quadrant][m-tile]
#pragma unroll
  for (int q = 0; q < 4; ++q)
#pragma unroll
    for (int i = 0; i < 2; ++i) acc[q][i] = f32x16{};

  const int l31 = l & 31, lhi = l >> 5;
  const int wmq = (w >> 2) * 64;
  const int wnq = (w & 3) * 32;

  auto phase = [&](int tile, int qa, int qb, int q, int wait6) {
    if (wait6 == 1)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if (wait6 == 0)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    const bf16* As = smem[tile & 1][0][qa];
    const bf16* Bs = smem[tile & 1][1][qb];
    bf16x8 af[2][4], bfr[4];
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      int chunk = kk * 2 + lhi;  // k = kk*16 + (l>>5)*8 .. +8
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
        af[mt][kk] = *(const bf16x8*)((const char*)As +
                                      t8_lds_byte(wmq + mt * 32 + l31,
                                                  chunk));
      bfr[kk] = *(const bf16x8*)((const char*)Bs +
                                 t8_lds_byte(wnq + l31, chunk));
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
        acc[q][mt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[mt][kk], bfr[kk], acc[q][mt], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  };

#pragma unroll
  for (int h = 0; h < 4; ++h) issue_half(h, 0);

  for (int tile = 0; tile < ksteps; ++tile) {
    int nxt = tile + 1;
    bool pre = nxt < ksteps;
    if (pre) issue_half(0, nxt);
    phase(tile, 0, 0, 0, pre ? 1 : 0);
    if (pre) issue_half(1, nxt);
    phase(tile, 1, 0, 1, pre ? 1 : -1);
    if (pre) issue_half(2, nxt);
    phase(tile, 1, 1, 2, pre ? 1 : -1);
    if (pre) issue_half(3, nxt);
    phase(tile, 0, 1, 3, -1);
  }

  bf16* cz = Cout + (long)z * strideC;
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    int qa = (q == 1 || q == 2) ? 1 : 0;
    int qb = (q >= 2) ? 1 : 0;
    int gcol = bn + qb * 128 + wnq + l31;
    if (gcol >= N) continue;
    float bv = BIAS ? bias[gcol] : 0.f;
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        int grow = bm + qa * 128 + wmq + mt * 32 + (reg & 3) +
                   8 * (reg >> 2) + 4 * lhi;
        if (grow >= M) continue;
        float v = acc[q][mt][reg] + bv;
        if (RELU) v = v > 0.f ? v : 0.f;
        cz[(long)grow * ldc + gcol] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Guide-faithful phase variant (§5 template discipline, MXA_TN8G):
// fixed 128x64 wave tiles (wave reads ONE A-half + ONE B-half only —
// halves the LDS read traffic of the quadrant-shared layout above),
// double-barrier phases ([ds_read; glds; s_barrier; lgkmcnt(0); setprio
// MFMA cluster; s_barrier]), ONE counted s_waitcnt vmcnt(4) per K-tile
// window, and a race-free issue schedule proved by read-confinement:
//   A frags fully read by p1 (A0/A1 regions rewritten by glds at p2/p3),
//   B frags read at p0/p2 (B regions rewritten at the NEXT window's p0,
//   which targets the other tile buffer).
// Window T (buffer T&1):  [vmcnt(4)]
//   p0: read a[0..3],b[0..1]; glds B0,B1 of T+1   ; MFMA mf0-3 x nf0-1
//   p1: read a[4..7]        ; glds  -             ; MFMA mf4-7 x nf0-1
//   p2: read b[2..3]        ; glds A0 of T+2      ; MFMA mf0-3 x nf2-3
//   p3:  -                  ; glds A1 of T+2      ; MFMA mf4-7 x nf2-3
// Tile T is fully resident at its window start: A halves issued at
// window T-2 (p2,p3), B halves at window T-1 (p0); vmcnt(4) leaves only
// tile T+1's A halves in flight.
// ---------------------------------------------------------------------------

template <bool RELU, bool BIAS>
__global__ __launch_bounds__(T8_THREADS, 1) void gemm_tn8g_core(
    Plain8Loader aload, Plain8Loader bload, bf16* __restrict__ Cout,
    const float* __restrict__ bias, int M, int N, long ldc, long strideC,
    int ksteps, long kstride) {
  __shared__ __align__(16) bf16 smem[2][2][2][128 * 64];

  const int ntn = (N + T8_BN - 1) / T8_BN;
  const int ntm = (M + T8_BM - 1) / T8_BM;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  // 4-row tile patches: an XCD's ~32 resident tiles form a 4(M) x 8(N)
  // patch (12 distinct A/B panels through its L2) instead of a row strip
  // (~18 panels) — the 256^2-tile GEMM is HBM-bound without this reuse.
  int tm, tnn;
  // patches only pay when blocks oversubscribe the CUs (multi-wave
  // grids); at <=1 block/CU the row-strip map measured faster (LLC)
  if ((ntm & 3) == 0 && ntm * ntn > 256) {
    tm = (bid / (4 * ntn)) * 4 + (bid & 3);
    tnn = (bid >> 2) % ntn;
  } else {
    tm = bid / ntn;
    tnn = bid % ntn;
  }
  const int bm = tm * T8_BM, bn = tnn * T8_BN;
  const int t = threadIdx.x;
  const int z = blockIdx.z;
  const int w = t / WAVE, l = t % WAVE;

  aload.init(bm, t, z);
  bload.init(bn, t, z);

  auto issue_half = [&](int op, int half, int tile) {
    long koff = (long)tile * kstride;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int p = half * 2 + j;
      unsigned seg = __builtin_amdgcn_readfirstlane(
          (unsigned)(w * 2048 + j * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smem[tile & 1][op][half]
              [0] + seg);
      const bf16* gp = op ? bload.src(p, koff) : aload.src(p, koff);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
          0, 0);
    }
  };

  const int lrow = l & 15, lk = l >> 4;
  const int wr = w >> 2;        // A half (M)
  const int wc = w & 3;         // N quarter
  const int bh = wc >> 1;       // B half
  const int bno = (wc & 1) * 64;  // N offset within the B half

  f32x4 acc[8][4];
#pragma unroll
  for (int mf = 0; mf < 8; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) acc[mf][nf] = f32x4{0.f, 0.f, 0.f, 0.f};

  bf16x8 a[8][2], b[4][2];

  auto read_a = [&](const bf16* As, int mf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      a[mf][kk] = *(const bf16x8*)((const char*)As +
                                   t8_lds_byte(mf * 16 + lrow, kk * 4 + lk));
  };
  auto read_b = [&](const bf16* Bs, int nf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      b[nf][kk] = *(const bf16x8*)((const char*)Bs +
                                   t8_lds_byte(bno + nf * 16 + lrow,
                                               kk * 4 + lk));
  };
  // one 16-MFMA cluster, no sync (the window body handles all sync)
  auto mma16 = [&](int mf0, int nf0) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[mf0 + mf][nf0 + nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mf0 + mf][kk], b[nf0 + nf][kk], acc[mf0 + mf][nf0 + nf], 0,
              0, 0);
  };

  // prologue: tile0 A0,A1,B0,B1 + tile1 A0,A1
  issue_half(0, 0, 0);
  issue_half(0, 1, 0);
  issue_half(1, 0, 0);
  issue_half(1, 1, 0);
  if (1 < ksteps) {
    issue_half(0, 0, 1);
    issue_half(0, 1, 1);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();  // publish prologue landings to all waves

  // Window body (2 barriers per K-tile): every fragment of the wave's
  // A-half and B-half is register-resident before any MFMA, so the 64
  // MFMAs run as one barrier-free setprio section.  Safety: the opening
  // lgkmcnt(0)+barrier publishes that all waves' LDS reads are DONE, so
  // the A-half glds of tile+2 (same buffer) issued inside the MFMA
  // section cannot overwrite data anyone still needs; the closing
  // vmcnt+barrier publishes every wave's glds landings for the next
  // window's reads.
  // Window body (2 barriers per K-tile): every fragment of the wave's
  // A-half and B-half is register-resident before any MFMA, so the 64
  // MFMAs run as one barrier-free setprio section.  Safety: the opening
  // lgkmcnt(0)+barrier publishes that all waves' LDS reads are DONE, so
  // the A-half glds of tile+2 (same buffer) issued inside the MFMA
  // section cannot overwrite data anyone still needs; the closing
  // vmcnt+barrier publishes every wave's glds landings for the next
  // window's reads.  (A deeper variant issuing all 4 halves of tile+2
  // here with vmcnt(8) measured 6-9% SLOWER — 16 loads in flight fight
  // the read zone for L2; B halves stay on the one-window-ahead path.)
  for (int tile = 0; tile < ksteps; ++tile) {
    const bf16* As = smem[tile & 1][0][wr];
    const bf16* Bs = smem[tile & 1][1][bh];
#pragma unroll
    for (int mf = 0; mf < 8; ++mf) read_a(As, mf);
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) read_b(Bs, nf);
    if (tile + 1 < ksteps) {
      issue_half(1, 0, tile + 1);
      issue_half(1, 1, tile + 1);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    mma16(0, 0);
    if (tile + 2 < ksteps) issue_half(0, 0, tile + 2);
    mma16(4, 0);
    if (tile + 2 < ksteps) issue_half(0, 1, tile + 2);
    mma16(0, 2);
    mma16(4, 2);
    __builtin_amdgcn_s_setprio(0);
    if (tile + 2 < ksteps)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (tile + 1 < ksteps)  // penultimate: drain tile+1 fully
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue
  bf16* cz = Cout + (long)z * strideC;
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int gcol = bn + wc * 64 + nf * 16 + (l & 15);
      if (gcol >= N) continue;
      float bv = BIAS ? bias[gcol] : 0.f;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + wr * 128 + mf * 16 + (l >> 4) * 4 + i;
        if (grow >= M) continue;
        float v = acc[mf][nf][i] + bv;
        if (RELU) v = v > 0.f ? v : 0.f;
        cz[(long)grow * ldc + gcol] = f2bf(v);
      }
    }
  }
}

// 32x32x16 window variant: identical sync/issue structure to
// gemm_tn8g_core but 32 x v_mfma_f32_32x32x16_bf16 per K-tile instead
// of 64 x 16x16x32 — half the MFMA issue slots for the same FLOPs (the
// aligned-structure PMC showed 46-48% SQ_WAIT_INST_ANY).  C/D layout:
// col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5).
template <bool RELU, bool BIAS>
__global__ __launch_bounds__(T8_THREADS, 1) void gemm_tn8g32_core(
    Plain8Loader aload, Plain8Loader bload, bf16* __restrict__ Cout,
    const float* __restrict__ bias, int M, int N, long ldc, long strideC,
    int ksteps, long kstride) {
  __shared__ __align__(16) bf16 smem[2][2][2][128 * 64];

  const int ntn = (N + T8_BN - 1) / T8_BN;
  const int ntm = (M + T8_BM - 1) / T8_BM;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm, tnn;
  if ((ntm & 3) == 0 && ntm * ntn > 256) {
    tm = (bid / (4 * ntn)) * 4 + (bid & 3);
    tnn = (bid >> 2) % ntn;
  } else {
    tm = bid / ntn;
    tnn = bid % ntn;
  }
  const int bm = tm * T8_BM, bn = tnn * T8_BN;
  const int t = threadIdx.x;
  const int z = blockIdx.z;
  const int w = t / WAVE, l = t % WAVE;

  aload.init(bm, t, z);
  bload.init(bn, t, z);

  auto issue_half = [&](int op, int half, int tile) {
    long koff = (long)tile * kstride;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int p = half * 2 + j;
      unsigned seg = __builtin_amdgcn_readfirstlane(
          (unsigned)(w * 2048 + j * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smem[tile & 1][op][half]
              [0] + seg);
      const bf16* gp = op ? bload.src(p, koff) : aload.src(p, koff);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
          0, 0);
    }
  };

  const int l31 = l & 31, lhi = l >> 5;
  const int wr = w >> 2;          // A half (M)
  const int wc = w & 3;           // N quarter
  const int bh = wc >> 1;         // B half
  const int bno = (wc & 1) * 64;  // N offset within the B half

  f32x16 acc[4][2];  // [m-frag 32][n-frag 32]
#pragma unroll
  for (int mf = 0; mf < 4; ++mf)
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) acc[mf][nf] = f32x16{};

  bf16x8 a[4][4], b[2][4];  // [frag][kstep of 16]

  auto read_a = [&](const bf16* As, int mf) {
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      a[mf][kk] = *(const bf16x8*)((const char*)As +
                                   t8_lds_byte(mf * 32 + l31,
                                               kk * 2 + lhi));
  };
  auto read_b = [&](const bf16* Bs, int nf) {
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      b[nf][kk] = *(const bf16x8*)((const char*)Bs +
                                   t8_lds_byte(bno + nf * 32 + l31,
                                               kk * 2 + lhi));
  };

  issue_half(0, 0, 0);
  issue_half(0, 1, 0);
  issue_half(1, 0, 0);
  issue_half(1, 1, 0);
  if (1 < ksteps) {
    issue_half(0, 0, 1);
    issue_half(0, 1, 1);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int tile = 0; tile < ksteps; ++tile) {
    const bf16* As = smem[tile & 1][0][wr];
    const bf16* Bs = smem[tile & 1][1][bh];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) read_a(As, mf);
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) read_b(Bs, nf);
    if (tile + 1 < ksteps) {
      issue_half(1, 0, tile + 1);
      issue_half(1, 1, tile + 1);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      if (kk == 1 && tile + 2 < ksteps) issue_half(0, 0, tile + 2);
      if (kk == 2 && tile + 2 < ksteps) issue_half(0, 1, tile + 2);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a[mf][kk], b[nf][kk], acc[mf][nf], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    if (tile + 2 < ksteps)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (tile + 1 < ksteps)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  bf16* cz = Cout + (long)z * strideC;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      int gcol = bn + wc * 64 + nf * 32 + l31;
      if (gcol >= N) continue;
      float bv = BIAS ? bias[gcol] : 0.f;
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        int grow = bm + wr * 128 + mf * 32 + (reg & 3) + 8 * (reg >> 2) +
                   4 * lhi;
        if (grow >= M) continue;
        float v = acc[mf][nf][reg] + bv;
        if (RELU) v = v > 0.f ? v : 0.f;
        cz[(long)grow * ldc + gcol] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Half-width window variant (256 x 128 tile): for short-K (K=768-class)
// grids the 256^2 kernel can't fill — halving BN doubles the tile count
// (8192x3072x768: 384 -> 768 blocks) while keeping the 2-barrier window
// discipline.  Waves are 4(M) x 2(N), 64x64 patches, acc[4][4]; B is a
// single [128][64] row-XOR image per buffer.
// ---------------------------------------------------------------------------

template <bool RELU, bool BIAS>
__global__ __launch_bounds__(T8_THREADS, 1) void gemm_tn8h_core(
    Plain8Loader aload, Plain8Loader bload, bf16* __restrict__ Cout,
    const float* __restrict__ bias, int M, int N, long ldc, long strideC,
    int ksteps, long kstride) {
  __shared__ __align__(16) bf16 smemA[2][2][128 * 64];
  __shared__ __align__(16) bf16 smemB[2][128 * 64];

  const int ntn = (N + 127) / 128;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = (bid / ntn) * T8_BM, bn = (bid % ntn) * 128;
  const int t = threadIdx.x;
  const int z = blockIdx.z;
  const int w = t / WAVE, l = t % WAVE;

  aload.init(bm, t, z);
  bload.init(bn, t, z);

  auto issue_a = [&](int half, int tile) {
    long koff = (long)tile * kstride;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int p = half * 2 + j;
      unsigned seg = __builtin_amdgcn_readfirstlane(
          (unsigned)(w * 2048 + j * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smemA[tile & 1][half][0] +
          seg);
      const bf16* gp = aload.src(p, koff);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
          0, 0);
    }
  };
  auto issue_b = [&](int tile) {  // B: single 128-row half (slots 0,1)
    long koff = (long)tile * kstride;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      unsigned seg = __builtin_amdgcn_readfirstlane(
          (unsigned)(w * 2048 + j * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smemB[tile & 1][0] + seg);
      const bf16* gp = bload.src(j, koff);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
          0, 0);
    }
  };

  const int lrow = l & 15, lk = l >> 4;
  const int wr2 = w >> 1;         // M quarter (64 rows)
  const int wc2 = w & 1;          // N half (64 cols)
  const int ah = wr2 >> 1;        // A half image
  const int amo = (wr2 & 1) * 64; // row offset within the half

  f32x4 acc[4][4];
#pragma unroll
  for (int mf = 0; mf < 4; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) acc[mf][nf] = f32x4{0.f, 0.f, 0.f, 0.f};

  bf16x8 a[4][2], b[4][2];

  // prologue: tile0 A+B, tile1 A (vmcnt(4) leaves tile1's A in flight)
  issue_a(0, 0);
  issue_a(1, 0);
  issue_b(0);
  if (1 < ksteps) {
    issue_a(0, 1);
    issue_a(1, 1);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int tile = 0; tile < ksteps; ++tile) {
    const bf16* As = smemA[tile & 1][ah];
    const bf16* Bs = smemB[tile & 1];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        a[mf][kk] = *(const bf16x8*)((const char*)As +
                                     t8_lds_byte(amo + mf * 16 + lrow,
                                                 kk * 4 + lk));
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        b[nf][kk] = *(const bf16x8*)((const char*)Bs +
                                     t8_lds_byte(wc2 * 64 + nf * 16 + lrow,
                                                 kk * 4 + lk));
    if (tile + 1 < ksteps) issue_b(tile + 1);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      if (kk == 0 && tile + 2 < ksteps) issue_a(0, tile + 2);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mf][kk], b[nf][kk], acc[mf][nf], 0, 0, 0);
      if (kk == 0 && tile + 2 < ksteps) issue_a(1, tile + 2);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 2; nf < 4; ++nf)
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

  bf16* cz = Cout + (long)z * strideC;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int gcol = bn + wc2 * 64 + nf * 16 + (l & 15);
      if (gcol >= N) continue;
      float bv = BIAS ? bias[gcol] : 0.f;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + ah * 128 + amo + mf * 16 + (l >> 4) * 4 + i;
        if (grow >= M) continue;
        float v = acc[mf][nf][i] + bv;
        if (RELU) v = v > 0.f ? v : 0.f;
        cz[(long)grow * ldc + gcol] = f2bf(v);
      }
    }
  }
}

static bool tn8_useg32() {
  static const bool v = [] {
    const char* e = getenv("MXA_TN8G32");
    return e && e[0] == '1';
  }();
  return v;
}

static bool tn8_useg() {
  // default ON (measured +12-24% vs gemm_tn8_core on every dense shape);
  // MXA_TN8G=0 falls back to the aligned-phase kernel
  static const bool v = [] {
    const char* e = getenv("MXA_TN8G");
    return !e || e[0] != '0';
  }();
  return v;
}

static bool tn8_use32() {
  static const bool v = [] {
    const char* e = getenv("MXA_TN8_32");
    return e && e[0] == '1';
  }();
  return v;
}

void gemm_tn8_bf16(const bf16* A, const bf16* B, bf16* C, const float* bias,
                   int M, int N, int K, long lda, long ldb, long ldc,
                   long strideA, long strideB, long strideC, int batches,
                   bool relu, const bf16* zpage, hipStream_t stream) {
  Plain8Loader a{A, zpage, M, lda, strideA};
  Plain8Loader b{B, zpage, N, ldb, strideB};
  dim3 grid(cdiv_h(M, T8_BM) * cdiv_h(N, T8_BN), 1, batches);
  int ksteps = K / T8_BK;
  if (tn8_useg32()) {
    if (bias) {
      if (relu)
        gemm_tn8g32_core<true, true><<<grid, T8_THREADS, 0, stream>>>(
            a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
      else
        gemm_tn8g32_core<false, true><<<grid, T8_THREADS, 0, stream>>>(
            a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    } else if (relu) {
      gemm_tn8g32_core<true, false><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    } else {
      gemm_tn8g32_core<false, false><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    }
    return;
  }
  if (tn8_useg()) {
    if (bias) {
      if (relu)
        gemm_tn8g_core<true, true><<<grid, T8_THREADS, 0, stream>>>(
            a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
      else
        gemm_tn8g_core<false, true><<<grid, T8_THREADS, 0, stream>>>(
            a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    } else if (relu) {
      gemm_tn8g_core<true, false><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    } else {
      gemm_tn8g_core<false, false><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    }
    return;
  }
  if (tn8_use32()) {
    if (bias) {
      if (relu)
        gemm_tn8x32_core<true, true><<<grid, T8_THREADS, 0, stream>>>(
            a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
      else
        gemm_tn8x32_core<false, true><<<grid, T8_THREADS, 0, stream>>>(
            a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    } else if (relu) {
      gemm_tn8x32_core<true, false><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    } else {
      gemm_tn8x32_core<false, false><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    }
    return;
  }
  if (bias) {
    if (relu)
      gemm_tn8_core<true, true><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    else
      gemm_tn8_core<false, true><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
  } else {
    if (relu)
      gemm_tn8_core<true, false><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    else
      gemm_tn8_core<false, false><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
  }
}

void gemm_tn8h_bf16(const bf16* A, const bf16* B, bf16* C,
                    const float* bias, int M, int N, int K, long lda,
                    long ldb, long ldc, long strideA, long strideB,
                    long strideC, int batches, bool relu,
                    const bf16* zpage, hipStream_t stream) {
  Plain8Loader a{A, zpage, M, lda, strideA};
  Plain8Loader b{B, zpage, N, ldb, strideB};
  dim3 grid(cdiv_h(M, T8_BM) * cdiv_h(N, 128), 1, batches);
  int ksteps = K / T8_BK;
  if (bias) {
    if (relu)
      gemm_tn8h_core<true, true><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
    else
      gemm_tn8h_core<false, true><<<grid, T8_THREADS, 0, stream>>>(
          a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
  } else if (relu) {
    gemm_tn8h_core<true, false><<<grid, T8_THREADS, 0, stream>>>(
        a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
  } else {
    gemm_tn8h_core<false, false><<<grid, T8_THREADS, 0, stream>>>(
        a, b, C, bias, M, N, ldc, strideC, ksteps, T8_BK);
  }
}

}  // namespace mxa
