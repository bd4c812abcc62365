// WG-form MFMA GEMM for gfx950: weight gradients (reduce over the MAJOR
// dimension of both operands):
//
//   C[Mo][No] += sum_m A[m][Mo] * B[m][No]
//
// Used for (reference parity: FullyConnectedGrad / conv wgrad in
// nn/fully_connected.cu, nn/convolution.cu):
//   * FullyConnected wgrad: dW[out][in] = sum_m dy[m][out] * x[m][in]
//   * Conv2D wgrad (NHWC) : dW[r,s,c,k] = sum_m xg[m][c] * dy[m][k]
//     where xg is the im2col gather; the (r,s) slice and the reduce-split
//     are encoded in blockIdx.y.
//
// Both operands are staged [m][cols] (coalesced 16-B row chunks) and
// written to LDS TRANSPOSED via in-register bf16 pair interleave (two
// m-rows zipped into b32 writes), producing a [cols][m] swizzled image —
// fragment reads are then plain ds_read_b128 rows exactly like the TN
// kernel (no transpose-read instruction).  The reduce dimension is split
// across blocks; f32 output accumulates with atomicAdd.
#include "common.h"

namespace mxa {

constexpr int WBM = 128, WBN = 128, WBK = 64, WTHREADS = 256;
constexpr int WT8 = 512;  // tr-variant thread count (8 waves per tile)

DEVINL int colkey(int col) { return (col ^ (col >> 3)) & 7; }

DEVINL int wg_byte(int col, int mbyte) {
  // byte offset of m-byte `mbyte` (0..127) in image column `col` of a
  // [128 cols][64 m] bf16 image; XOR swizzle on the 16-B chunk index.
  return col * 128 + ((((mbyte >> 4) ^ colkey(col)) << 4) | (mbyte & 15));
}

struct WgPlainRows {
  const bf16* base;
  const bf16* zpage;
  long mtotal;
  long ld;
  int rowlen;  // valid elements per row (col tail guard; must be %8)
  // device state
  long cur;
  int col0;

  DEVINL void init(int c0, long m0) {
    col0 = c0;
    cur = m0;
  }
  DEVINL void step() { cur += WBK; }
  DEVINL const bf16* src(int mrow, int colchunk) const {
    long m = cur + mrow;
    int c = col0 + colchunk * 8;
    return (m < mtotal && c + 8 <= rowlen)
               ? base + m * ld + c
               : zpage + colchunk * 8;
  }
};

struct WgIm2colRows {
  const bf16* x;
  const bf16* zpage;
  int Nb, H, W, C, P, Q;
  int sh, sw, ph, pw;
  // device state
  int roff, soff;
  int n_[4], p_[4], q_[4];
  int col0;

  DEVINL void init_rows(int c0, long m0, const int* rows) {
    col0 = c0;
    int PQ = P * Q;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      long m = m0 + rows[i];
      n_[i] = (int)(m / PQ);
      int pq = (int)(m % PQ);
      p_[i] = pq / Q;
      q_[i] = pq % Q;
    }
  }
  DEVINL void step() {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      q_[i] += WBK;
      while (q_[i] >= Q) {
        q_[i] -= Q;
        p_[i] += 1;
      }
      while (p_[i] >= P) {
        p_[i] -= P;
        n_[i] += 1;
      }
    }
  }
  DEVINL const bf16* src(int slot, int colchunk) const {
    int c = col0 + colchunk * 8;
    int n = n_[slot];
    // column tail (tile wider than C, e.g. C=64 under a 128-wide tile)
    // must read the zero page — past-row reads can cross the allocation.
    if (n >= Nb || c + 8 > C) return zpage + colchunk * 8;
    int ih = p_[slot] * sh - ph + roff;
    int iw = q_[slot] * sw - pw + soff;
    if (ih < 0 || ih >= H || iw < 0 || iw >= W) return zpage + colchunk * 8;
    return x + (((long)n * H + ih) * W + iw) * C + c;
  }
};

template <bool A_IM2COL>
__global__ __launch_bounds__(WTHREADS, 2) void gemm_wg_core(
    WgPlainRows aplain, WgIm2colRows aim, WgPlainRows bload,
    float* __restrict__ Cout, int Mo, int No, long ldc, long strideSeg,
    int S, int dh, int dw, int ksteps_total, int nsplit, int use_atomic,
    long strideSlot) {
  __shared__ __align__(16) bf16 smem[2][2][WBM * WBK];

  const int seg = blockIdx.y / nsplit, split = blockIdx.y % nsplit;
  const int per = (ksteps_total + nsplit - 1) / nsplit;
  const int kstep0 = split * per;
  const int ksteps = min(per, ksteps_total - kstep0);
  if (ksteps <= 0) return;

  const int ntn = (No + WBN - 1) / WBN;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = (bid / ntn) * WBM, bn = (bid % ntn) * WBN;
  const int t = threadIdx.x;
  // Balanced staging: EVERY thread stages half of A and half of B (a
  // wave that staged only the contiguous operand would park at the
  // barrier while the gather waves straggle — measured 41% SQ_WAIT_ANY).
  // Per thread: 2 row-pairs per operand; pair = t&15, colchunk = t>>4.
  const int pair = t & 15, colchunk = t >> 4;  // colchunk 0..15

  int rows_local[4];
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    rows_local[2 * p] = (p * 16 + pair) * 2;
    rows_local[2 * p + 1] = (p * 16 + pair) * 2 + 1;
  }

  long m0 = (long)kstep0 * WBK;
  if (A_IM2COL) {
    aim.roff = (seg / S) * dh;
    aim.soff = (seg % S) * dw;
    aim.init_rows(bm, m0, rows_local);
  } else {
    aplain.init(bm, m0);
  }
  bload.init(bn, m0);

  i32x4 sregA[4], sregB[4];

  auto stage_load = [&]() {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (A_IM2COL)
        sregA[i] = *(const i32x4*)aim.src(i, colchunk);
      else
        sregA[i] = *(const i32x4*)aplain.src(rows_local[i], colchunk);
      sregB[i] = *(const i32x4*)bload.src(rows_local[i], colchunk);
    }
    if (A_IM2COL)
      aim.step();
    else
      aplain.step();
    bload.step();
  };

  auto stage_write = [&](int buf) {
#pragma unroll
    for (int op = 0; op < 2; ++op) {
      bf16* img = smem[buf][op];
      const i32x4* sreg = op ? sregB : sregA;
#pragma unroll
      for (int p = 0; p < 2; ++p) {
        const u32* lo = (const u32*)&sreg[2 * p];      // row r0, 8 bf16
        const u32* hi = (const u32*)&sreg[2 * p + 1];  // row r0+1
        int r0 = rows_local[2 * p];
#pragma unroll
        for (int jw = 0; jw < 4; ++jw) {
          u32 a01 = lo[jw];  // cols (2jw, 2jw+1) of row r0
          u32 b01 = hi[jw];
          u32 pair0 = (a01 & 0xffffu) | (b01 << 16);
          u32 pair1 = (a01 >> 16) | (b01 & 0xffff0000u);
          int c0 = colchunk * 8 + 2 * jw;
          *(u32*)((char*)img + wg_byte(c0, r0 * 2)) = pair0;
          *(u32*)((char*)img + wg_byte(c0 + 1, r0 * 2)) = pair1;
        }
      }
    }
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int w = t / WAVE, l = t % WAVE;
  const int wm = (w >> 1) * 64, wn = (w & 1) * 64;
  const int lcol = l & 15, lk = l >> 4;

  auto compute = [&](int buf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[4], bfr[4];
      int mbyte = (kk * 4 + lk) << 4;
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) {
        int col = wm + mf * 16 + lcol;
        af[mf] =
            *(const bf16x8*)((const char*)smem[buf][0] + wg_byte(col, mbyte));
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int col = wn + nf * 16 + lcol;
        bfr[nf] =
            *(const bf16x8*)((const char*)smem[buf][1] + wg_byte(col, mbyte));
      }
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
    }
  };

  stage_load();
  stage_write(0);
  __syncthreads();
  if (ksteps > 1) stage_load();
  for (int ks = 0; ks < ksteps; ++ks) {
    int buf = ks & 1;
    compute(buf);
    if (ks + 1 < ksteps) {
      stage_write(buf ^ 1);
      if (ks + 2 < ksteps) stage_load();
    }
    __syncthreads();
  }

  // slot-buffer epilogue: each split owns a private f32 slot (plain
  // stores, no HBM atomics, no pre-zero memset); wg_fold sums the slots
  float* cseg = Cout + (long)seg * strideSeg + (long)split * strideSlot;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int gcol = bn + wn + nf * 16 + (l & 15);
      if (gcol >= No) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + wm + mf * 16 + (l >> 4) * 4 + i;
        if (grow >= Mo) continue;
        float v = acc[mf][nf][i];
        if (use_atomic)
          atomicAdd(&cseg[(long)grow * ldc + gcol], v);
        else
          cseg[(long)grow * ldc + gcol] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// tr-variant: hardware-transpose fragment reads (ds_read_b64_tr_b16)
// ---------------------------------------------------------------------------
// Empirically probed semantics (tools/probe_tr.hip on gfx950): within each
// 16-lane group, writing the in-group lane index as (r,c) = (l>>2, l&3),
// output slot j of lane (r*4+c) is slot c of the 8-byte segment addressed
// by lane (j*4+r) — a rotation of the (lane_hi2, lane_lo2, slot) cube.
// Consequence: tiles can be staged ROW-major [m][cols] straight from
// global memory with global_load_lds (no in-register zip, no LDS write
// instructions at all), and MFMA k-major fragments come out of two tr
// reads per lane.  The zip kernel above is kept as a fallback
// (MXA_WG_ZIP=1) and an A/B baseline.
//
// LDS image per operand: [64 m-rows][128 cols] bf16, 256-B rows, chunk16
// XOR swizzle  chunk' = chunk ^ ((row&3)<<1)  so the 4 rows of one tr
// read land in disjoint 32-B bank windows (even bank coverage).

DEVINL int tr_byte(int row, int cbyte) {
  return row * 256 + ((((cbyte >> 4) ^ ((row & 3) << 1)) << 4) | (cbyte & 15));
}

struct TrPlainRows {
  const bf16* base;
  const bf16* zpage;
  long mtotal;
  long ld;
  int rowlen;
  // device state
  const bf16* ptr[2];  // pass p -> this thread's source row pointer
  long mrem[2];
  bool colok;
  int cgbyte;

  DEVINL void init(int col0, long m0, const int* myrow, int cg) {
    cgbyte = cg * 16;
    int col = col0 + cg * 8;
    colok = col + 8 <= rowlen;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      ptr[p] = base + (m0 + myrow[p]) * ld + col;
      mrem[p] = mtotal - (m0 + myrow[p]);
    }
  }
  DEVINL void step() {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      ptr[p] += (long)WBK * ld;
      mrem[p] -= WBK;
    }
  }
  DEVINL const bf16* src(int p) const {
    return (colok && mrem[p] > 0) ? ptr[p]
                                  : (const bf16*)((const char*)zpage + cgbyte);
  }

  // 4-slot variant for the 256-wide w64 A image: per-slot column chunk
  // (the tr512 swizzle key varies by row) and per-slot colok
  const bf16* ptr4[4];
  long mrem4[4];
  bool colok4[4];
  int cgbyte4[4];
  DEVINL void init4(int col0, long m0, const int* rows, const int* cg) {
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      cgbyte4[p] = (cg[p] & 15) * 16;
      int col = col0 + cg[p] * 8;
      colok4[p] = col + 8 <= rowlen;
      ptr4[p] = base + (m0 + rows[p]) * ld + col;
      mrem4[p] = mtotal - (m0 + rows[p]);
    }
  }
  DEVINL const bf16* src4(int p) const {
    return (colok4[p] && mrem4[p] > 0)
               ? ptr4[p]
               : (const bf16*)((const char*)zpage + cgbyte4[p]);
  }
  DEVINL void step4() {
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      ptr4[p] += (long)WBK * ld;
      mrem4[p] -= WBK;
    }
  }
};

struct TrIm2colRows {
  const bf16* x;
  const bf16* zpage;
  int Nb, H, W, C, P, Q;
  int sh, sw, ph, pw;
  // device state
  int roff, soff;
  int n_[2], p_[2], q_[2];
  bool colok;
  int col, cgbyte;

  DEVINL void init(int col0, long m0, const int* myrow, int cg) {
    cgbyte = cg * 16;
    col = col0 + cg * 8;
    colok = col + 8 <= C;
    int PQ = P * Q;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      long m = m0 + myrow[p];
      n_[p] = (int)(m / PQ);
      int pq = (int)(m % PQ);
      p_[p] = pq / Q;
      q_[p] = pq % Q;
    }
  }
  DEVINL void step() {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      q_[i] += WBK;
      while (q_[i] >= Q) {
        q_[i] -= Q;
        p_[i] += 1;
      }
      while (p_[i] >= P) {
        p_[i] -= P;
        n_[i] += 1;
      }
    }
  }
  DEVINL const bf16* src(int p) const {
    const bf16* zp = (const bf16*)((const char*)zpage + cgbyte);
    if (n_[p] >= Nb || !colok) return zp;
    int ih = p_[p] * sh - ph + roff;
    int iw = q_[p] * sw - pw + soff;
    if (ih < 0 || ih >= H || iw < 0 || iw >= W) return zp;
    return x + (((long)n_[p] * H + ih) * W + iw) * C + col;
  }
};

template <bool A_IM2COL, bool OUTBF16 = false>
__global__ __launch_bounds__(WT8, 2) void gemm_wg_tr_core(
    TrPlainRows aplain, TrIm2colRows aim, TrPlainRows bload,
    float* __restrict__ Cout, int Mo, int No, long ldc, long strideSeg,
    int S, int dh, int dw, int ksteps_total, int nsplit, int use_atomic,
    ZMap za, ZMap zb, ZMap zc, long strideSlot = 0,
    float* __restrict__ biasg = nullptr) {
  __shared__ __align__(16) bf16 smem[2][2][WBK * WBM];  // [buf][op][64][128]

  const int zb_ = blockIdx.z;
  if (A_IM2COL)
    aim.x += za.off(zb_);
  else
    aplain.base += za.off(zb_);
  bload.base += zb.off(zb_);
  // zc.off() counts OUTPUT ELEMENTS: 2 bytes in bf16 mode, 4 in f32
  Cout = (float*)((char*)Cout + zc.off(zb_) * (OUTBF16 ? 2 : 4));

  const int seg = blockIdx.y / nsplit, split = blockIdx.y % nsplit;
  const int per = (ksteps_total + nsplit - 1) / nsplit;
  const int kstep0 = split * per;
  const int ksteps = min(per, ksteps_total - kstep0);
  if (ksteps <= 0) return;

  const int ntn = (No + WBN - 1) / WBN;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = (bid / ntn) * WBM, bn = (bid % ntn) * WBN;
  const int t = threadIdx.x;
  const int lw = t / WAVE, ll = t % WAVE;

  // glds slot of pass p (8 waves x 2 passes): LDS bytes
  // (lw*2+p)*1024 + ll*16 -> row = lw*8 + p*4 + (ll>>4), chunk ll&15.
  // Swizzle moves to the SOURCE chunk: cg = (ll&15) ^ key(row),
  // key(row) = (row&3)<<1 = (ll>>4)<<1 (uniform per lane).
  const int cg = (ll & 15) ^ ((ll >> 4) << 1);
  int myrow[2];
#pragma unroll
  for (int p = 0; p < 2; ++p) myrow[p] = lw * 8 + p * 4 + (ll >> 4);

  long m0 = (long)kstep0 * WBK;
  if (A_IM2COL) {
    aim.roff = (seg / S) * dh;
    aim.soff = (seg % S) * dw;
    aim.init(bm, m0, myrow, cg);
  } else {
    aplain.init(bm, m0, myrow, cg);
  }
  bload.init(bn, m0, myrow, cg);

  auto stage_glds = [&](int buf) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      unsigned lbase =
          __builtin_amdgcn_readfirstlane((unsigned)((lw * 2 + p) * 1024));
      auto ldsA = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smem[buf][0][0] + lbase);
      auto ldsB = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smem[buf][1][0] + lbase);
      auto gA = (const __attribute__((address_space(1))) unsigned int*)(
          A_IM2COL ? aim.src(p) : aplain.src(p));
      auto gB =
          (const __attribute__((address_space(1))) unsigned int*)bload.src(p);
      __builtin_amdgcn_global_load_lds(gA, ldsA, 16, 0, 0);
      __builtin_amdgcn_global_load_lds(gB, ldsB, 16, 0, 0);
    }
    if (A_IM2COL)
      aim.step();
    else
      aplain.step();
    bload.step();
  };

  f32x4 acc[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // 8 waves as 4(Mo) x 2(No): each wave a 32x64 patch
  const int wm = (lw & 3) * 32, wn = (lw >> 2) * 64;
  const int ig = ll & 15, khi = ll >> 4;

  // fused bias gradient (A = dy): blocks in the first No-tile column of
  // seg 0 also fold their A fragments over the reduce dim — kills the
  // standalone colsum pass over dy (the A tile is already in registers)
  // only the wn==0 wave group folds (each mo column is read by BOTH
  // N-halves' waves; counting one avoids the 2x)
  const bool do_bias = biasg != nullptr && bn == 0 && seg == 0 &&
                       (lw >> 2) == 0;
  float bsum[2] = {0.f, 0.f};

  auto trread = [&](const bf16* img, int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)img + tr_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      // fragment k-rows this lane addresses: kk*32 + khi*8 + h*4 + (ig>>2)
      int rbase = kk * 32 + khi * 8 + (ig >> 2);
      int cb = (ig & 3) * 8;
      bf16x8 af[2], bfr[4];
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
        int cbyte = (wm + mf * 16) * 2 + cb;
        bf16x4 lo = trread(smem[buf][0], rbase, cbyte);
        bf16x4 hi = trread(smem[buf][0], rbase + 4, cbyte);
        af[mf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
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
      if (do_bias) {
#pragma unroll
        for (int mf = 0; mf < 2; ++mf)
#pragma unroll
          for (int j = 0; j < 8; ++j) bsum[mf] += bf2f(af[mf][j]);
      }
    }
  };

  // 2-phase glds schedule (same as gemm_tn): issue next-tile DMA before
  // computing the current tile; __syncthreads() drains the DMA.
  stage_glds(0);
  __syncthreads();
  for (int ks = 0; ks < ksteps; ++ks) {
    int buf = ks & 1;
    if (ks + 1 < ksteps) stage_glds(buf ^ 1);
    compute(buf);
    __syncthreads();
  }

  float* cseg = Cout + (long)seg * strideSeg + (long)split * strideSlot;
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int gcol = bn + wn + nf * 16 + (ll & 15);
      if (gcol >= No) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + wm + mf * 16 + (ll >> 4) * 4 + i;
        if (grow >= Mo) continue;
        float v = acc[mf][nf][i];
        if (OUTBF16)
          ((bf16*)cseg)[(long)grow * ldc + gcol] = f2bf(v);
        else if (use_atomic)
          atomicAdd(&cseg[(long)grow * ldc + gcol], v);
        else
          cseg[(long)grow * ldc + gcol] = v;
      }
    }
  }

  if (do_bias) {
    // A-fragment lane layout: column = wm + mf*16 + (ll&15); the four
    // ll>>4 groups hold disjoint reduce slices — fold then one atomic
    // per column per block (splits sum via the atomic)
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
      float v = bsum[mf];
      v += __shfl_xor(v, 16, 64);
      v += __shfl_xor(v, 32, 64);
      int col = bm + wm + mf * 16 + (ll & 15);
      if ((ll >> 4) == 0 && col < Mo) atomicAdd(&biasg[col], v);
    }
  }
}

// ---------------------------------------------------------------------------
// 64x64-wave-patch tr-variant (linear wgrad only, MXA_WG64): fragment
// reads scale with mf+nf while MFMAs scale with mf*nf, so widening the
// wave patch from 32x64 (16 MFMA : 12 tr-pairs per K-step) to 64x64
// (32 MFMA : 16 tr-pairs) raises MFMA per LDS instruction by 50% — the
// tr16 ISSUE rate is what bounds the 2-operand out-contiguous wgrad
// (profiles/RESULTS.md r3 lead #2).  Tile 256(Mo) x 128(No), 8 waves as
// 4x2; A image [64 m][256 Mo] with a 4-bit tr-swizzle key so the
// 16-row read sets spread across banks; B keeps the [64][128] format.
// ---------------------------------------------------------------------------

DEVINL int tr512_byte(int row, int cbyte) {
  int key = ((row & 3) << 2) ^ ((row >> 3) & 3);
  return row * 512 + ((((cbyte >> 4) ^ key) << 4) | (cbyte & 15));
}

template <bool OUTBF16>
__global__ __launch_bounds__(WT8, 1) void gemm_wg_w64_core(
    TrPlainRows aload, TrPlainRows bload, float* __restrict__ Cout,
    int Mo, int No, long ldc, int ksteps_total, int nsplit,
    long strideSlot, float* __restrict__ biasg) {
  __shared__ __align__(16) bf16 smemA[2][64 * 256];
  __shared__ __align__(16) bf16 smemB[2][64 * 128];

  const int ntn = (No + 127) / 128;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = (bid / ntn) * 256, bn = (bid % ntn) * 128;
  const int t = threadIdx.x;
  const int lw = t / WAVE, ll = t % WAVE;

  const int split = blockIdx.y;
  const int per = (ksteps_total + nsplit - 1) / nsplit;
  const int kstep0 = split * per;
  const int ksteps = min(per, ksteps_total - kstep0);
  if (ksteps <= 0) return;

  // A staging: 4 glds passes of 16 rows x 512 B; pass p covers rows
  // p*16 + (t>>5), 16-B chunk (t&31), source chunk XOR'd with the
  // 4-bit key so the LDS image matches tr512_byte
  int arow[4], acg[4];
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    arow[p] = p * 16 + (t >> 5);
    int key = ((arow[p] & 3) << 2) ^ ((arow[p] >> 3) & 3);
    acg[p] = (t & 31) ^ key;
  }
  // B staging: 2 passes of the existing [64][128] geometry
  int brow[2];
#pragma unroll
  for (int p = 0; p < 2; ++p) brow[p] = lw * 8 + p * 4 + (ll >> 4);
  const int bcg = (ll & 15) ^ ((ll >> 4) << 1);

  long m0 = (long)kstep0 * WBK;
  aload.init4(bm, m0, arow, acg);
  bload.init(bn, m0, brow, bcg);

  auto stage_glds = [&](int buf) {
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      unsigned lbase = __builtin_amdgcn_readfirstlane(
          (unsigned)((p * 16 + (lw * 2)) * 512));
      auto ldsA = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smemA[buf][0] + lbase);
      auto gA =
          (const __attribute__((address_space(1))) unsigned int*)aload.src4(
              p);
      __builtin_amdgcn_global_load_lds(gA, ldsA, 16, 0, 0);
    }
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      unsigned lbase =
          __builtin_amdgcn_readfirstlane((unsigned)((lw * 2 + p) * 1024));
      auto ldsB = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&smemB[buf][0] + lbase);
      auto gB =
          (const __attribute__((address_space(1))) unsigned int*)bload.src(p);
      __builtin_amdgcn_global_load_lds(gB, ldsB, 16, 0, 0);
    }
    aload.step4();
    bload.step();
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int wm = (lw & 3) * 64, wn = (lw >> 2) * 64;
  const int ig = ll & 15, khi = ll >> 4;

  // wn==0 wave group only: both N-halves' waves read every mo column
  const bool do_bias = biasg != nullptr && bn == 0 && (lw >> 2) == 0;
  float bsum[4] = {0.f, 0.f, 0.f, 0.f};

  auto trreadA = [&](const bf16* img, int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)img +
        tr512_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };
  auto trreadB = [&](const bf16* img, int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)img + tr_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int rbase = kk * 32 + khi * 8 + (ig >> 2);
      int cb = (ig & 3) * 8;
      bf16x8 af[4], bfr[4];
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) {
        int cbyte = (wm + mf * 16) * 2 + cb;
        bf16x4 lo = trreadA(smemA[buf], rbase, cbyte);
        bf16x4 hi = trreadA(smemA[buf], rbase + 4, cbyte);
        af[mf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int cbyte = (wn + nf * 16) * 2 + cb;
        bf16x4 lo = trreadB(smemB[buf], rbase, cbyte);
        bf16x4 hi = trreadB(smemB[buf], rbase + 4, cbyte);
        bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (do_bias) {
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
#pragma unroll
          for (int j = 0; j < 8; ++j) bsum[mf] += bf2f(af[mf][j]);
      }
    }
  };

  stage_glds(0);
  __syncthreads();
  for (int ks = 0; ks < ksteps; ++ks) {
    int buf = ks & 1;
    if (ks + 1 < ksteps) stage_glds(buf ^ 1);
    compute(buf);
    __syncthreads();
  }

  float* cseg = Cout + (long)split * strideSlot;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int gcol = bn + wn + nf * 16 + (ll & 15);
      if (gcol >= No) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + wm + mf * 16 + (ll >> 4) * 4 + i;
        if (grow >= Mo) continue;
        float v = acc[mf][nf][i];
        if (OUTBF16)
          ((bf16*)cseg)[(long)grow * ldc + gcol] = f2bf(v);
        else
          cseg[(long)grow * ldc + gcol] = v;
      }
    }
  }

  if (do_bias) {
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
      float v = bsum[mf];
      v += __shfl_xor(v, 16, 64);
      v += __shfl_xor(v, 32, 64);
      int col = bm + wm + mf * 16 + (ll & 15);
      if ((ll >> 4) == 0 && col < Mo) atomicAdd(&biasg[col], v);
    }
  }
}

// ---------------------------------------------------------------------------
// Ring-pipelined tr-variant: 32-row half-tiles on a counted-vmcnt glds ring
// ---------------------------------------------------------------------------
// The 2-phase schedule above keeps only ONE 64-row tile in flight while the
// per-tile MFMA work is short — rocprofv3 PMC showed the wgrad kernels
// 25% MFMA-busy / 41-52% wave-parked (profiles/RESULTS.md).  Same cure as
// gemm_tn8.hip: half the staging quantum (32 k-rows = 1 glds per operand
// per lane), put RING-1 halves in flight, and wait with a COUNTED
// s_waitcnt vmcnt across raw s_barriers so the wait never drains the
// younger prefetches.  RING=4 keeps LDS at 64 KiB (2 blocks/CU); RING=8
// matches tn8's depth at 1 block/CU.

struct TrPlainRows1 {
  const bf16* base;
  const bf16* zpage;
  long mtotal;
  long ld;
  int rowlen;
  // device state
  const bf16* ptr;
  long mrem;
  bool colok;
  int cgbyte;

  DEVINL void init(int col0, long m0, int myrow, int cg) {
    cgbyte = cg * 16;
    int col = col0 + cg * 8;
    colok = col + 8 <= rowlen;
    ptr = base + (m0 + myrow) * ld + col;
    mrem = mtotal - (m0 + myrow);
  }
  DEVINL void step() {
    ptr += 32 * ld;
    mrem -= 32;
  }
  DEVINL const bf16* src() const {
    return (colok && mrem > 0) ? ptr
                               : (const bf16*)((const char*)zpage + cgbyte);
  }
};

struct TrIm2colRows1 {
  const bf16* x;
  const bf16* zpage;
  int Nb, H, W, C, P, Q;
  int sh, sw, ph, pw;
  // device state
  int roff, soff;
  int n_, p_, q_;
  bool colok;
  int col, cgbyte;

  DEVINL void init(int col0, long m0, int myrow, int cg) {
    cgbyte = cg * 16;
    col = col0 + cg * 8;
    colok = col + 8 <= C;
    int PQ = P * Q;
    long m = m0 + myrow;
    n_ = (int)(m / PQ);
    int pq = (int)(m % PQ);
    p_ = pq / Q;
    q_ = pq % Q;
  }
  DEVINL void step() {
    q_ += 32;
    while (q_ >= Q) {
      q_ -= Q;
      p_ += 1;
    }
    while (p_ >= P) {
      p_ -= P;
      n_ += 1;
    }
  }
  DEVINL const bf16* src() const {
    const bf16* zp = (const bf16*)((const char*)zpage + cgbyte);
    if (n_ >= Nb || !colok) return zp;
    int ih = p_ * sh - ph + roff;
    int iw = q_ * sw - pw + soff;
    if (ih < 0 || ih >= H || iw < 0 || iw >= W) return zp;
    return x + (((long)n_ * H + ih) * W + iw) * C + col;
  }
};

// counted wait: `rem` halves still wanted in flight after this wait
// (2 glds per half); immediates only, so a small switch.
DEVINL void wg_ring_wait(int rem) {
  switch (rem) {
    case 0: asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); break;
    case 1: asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); break;
    case 2: asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); break;
    case 3: asm volatile("s_waitcnt vmcnt(6)" ::: "memory"); break;
    case 4: asm volatile("s_waitcnt vmcnt(8)" ::: "memory"); break;
    case 5: asm volatile("s_waitcnt vmcnt(10)" ::: "memory"); break;
    case 6: asm volatile("s_waitcnt vmcnt(12)" ::: "memory"); break;
    default: asm volatile("s_waitcnt vmcnt(14)" ::: "memory"); break;
  }
}

template <bool A_IM2COL, bool OUTBF16, int RING>
__global__ __launch_bounds__(WT8, RING <= 4 ? 2 : 1) void gemm_wg_ring_core(
    TrPlainRows1 aplain, TrIm2colRows1 aim, TrPlainRows1 bload,
    float* __restrict__ Cout, int Mo, int No, long ldc, long strideSeg,
    int S, int dh, int dw, int hsteps_total, int nsplit, int use_atomic,
    ZMap za, ZMap zb, ZMap zc, long strideSlot = 0) {
  // [ring slot][op][32 rows][128 cols]
  __shared__ __align__(16) bf16 smem[RING][2][32 * 128];

  const int zb_ = blockIdx.z;
  if (A_IM2COL)
    aim.x += za.off(zb_);
  else
    aplain.base += za.off(zb_);
  bload.base += zb.off(zb_);
  Cout = (float*)((char*)Cout + zc.off(zb_) * (OUTBF16 ? 2 : 4));

  const int seg = blockIdx.y / nsplit, split = blockIdx.y % nsplit;
  const int per = (hsteps_total + nsplit - 1) / nsplit;
  const int h0 = split * per;
  const int nh = min(per, hsteps_total - h0);
  if (nh <= 0) return;

  const int ntn = (No + WBN - 1) / WBN;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int bm = (bid / ntn) * WBM, bn = (bid % ntn) * WBN;
  const int t = threadIdx.x;
  const int lw = t / WAVE, ll = t % WAVE;

  // glds slot (8 waves x 32 rows): LDS bytes lw*1024 + ll*16 ->
  // row = lw*4 + (ll>>4), chunk ll&15; swizzle on the SOURCE chunk.
  const int cg = (ll & 15) ^ ((ll >> 4) << 1);
  const int myrow = lw * 4 + (ll >> 4);

  long m0 = (long)h0 * 32;
  if (A_IM2COL) {
    aim.roff = (seg / S) * dh;
    aim.soff = (seg % S) * dw;
    aim.init(bm, m0, myrow, cg);
  } else {
    aplain.init(bm, m0, myrow, cg);
  }
  bload.init(bn, m0, myrow, cg);

  auto issue_half = [&](int slot) {
    unsigned lbase = __builtin_amdgcn_readfirstlane((unsigned)(lw * 1024));
    auto ldsA = (__attribute__((address_space(3))) unsigned int*)(
        (__attribute__((address_space(3))) char*)&smem[slot][0][0] + lbase);
    auto ldsB = (__attribute__((address_space(3))) unsigned int*)(
        (__attribute__((address_space(3))) char*)&smem[slot][1][0] + lbase);
    auto gA = (const __attribute__((address_space(1))) unsigned int*)(
        A_IM2COL ? aim.src() : aplain.src());
    auto gB =
        (const __attribute__((address_space(1))) unsigned int*)bload.src();
    __builtin_amdgcn_global_load_lds(gA, ldsA, 16, 0, 0);
    __builtin_amdgcn_global_load_lds(gB, ldsB, 16, 0, 0);
    if (A_IM2COL)
      aim.step();
    else
      aplain.step();
    bload.step();
  };

  f32x4 acc[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int wm = (lw & 3) * 32, wn = (lw >> 2) * 64;
  const int ig = ll & 15, khi = ll >> 4;

  auto trread = [&](const bf16* img, int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)img + tr_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };

  auto compute_half = [&](int slot) {
    int rbase = khi * 8 + (ig >> 2);
    int cb = (ig & 3) * 8;
    bf16x8 af[2], bfr[4];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
      int cbyte = (wm + mf * 16) * 2 + cb;
      bf16x4 lo = trread(smem[slot][0], rbase, cbyte);
      bf16x4 hi = trread(smem[slot][0], rbase + 4, cbyte);
      af[mf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
    }
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int cbyte = (wn + nf * 16) * 2 + cb;
      bf16x4 lo = trread(smem[slot][1], rbase, cbyte);
      bf16x4 hi = trread(smem[slot][1], rbase + 4, cbyte);
      bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

  // prologue: RING-1 halves in flight
#pragma unroll
  for (int i = 0; i < RING - 1; ++i)
    if (i < nh) issue_half(i);

  for (int h = 0; h < nh; ++h) {
    int ahead = nh - 1 - h;  // younger halves wanted in flight
    if (ahead >= RING - 1) issue_half((h + RING - 1) % RING);
    wg_ring_wait(min(ahead, RING - 1));
    __builtin_amdgcn_s_barrier();
    compute_half(h % RING);
    __builtin_amdgcn_s_barrier();
  }

  float* cseg = Cout + (long)seg * strideSeg + (long)split * strideSlot;
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int gcol = bn + wn + nf * 16 + (ll & 15);
      if (gcol >= No) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int grow = bm + wm + mf * 16 + (ll >> 4) * 4 + i;
        if (grow >= Mo) continue;
        float v = acc[mf][nf][i];
        if (OUTBF16)
          ((bf16*)cseg)[(long)grow * ldc + gcol] = f2bf(v);
        else if (use_atomic)
          atomicAdd(&cseg[(long)grow * ldc + gcol], v);
        else
          cseg[(long)grow * ldc + gcol] = v;
      }
    }
  }

}

// ---------------------------------------------------------------------------
// Seg-merged 3x3 stride-1 conv wgrad ("c9"): all 9 (r,s) tiles per block
// ---------------------------------------------------------------------------
// The per-seg wgrad above re-reads x AND dy once per (r,s) — 9x the HBM
// traffic — and the round-1 PMC evidence showed the 3x3 wgrads
// staging-bound (25% MFMA busy).  For the same-size stride-1 case
// (P==H, Q==W, pad 1, dilation 1 — every ResNet 3x3) there is a clean
// restructuring: walk the reduce dimension m' over the PADDED output grid
// (P+2)x(Q+2).  Then
//   * dy rows are a gather with zeros at the padding positions (read ONCE),
//   * the im2col of a 1-padded COPY of x becomes a pure LINEAR stream:
//     xp_linear(m', r, s) = m' + (r-1)*(W+2) + (s-1)  — every (r,s) is a
//     row-SHIFT of one staged stream, so x is staged ONCE per k-step into
//     a 4-slot circular LDS window and all 9 accumulators are fed from it.
//   * every x row that falls outside a seg's valid window pairs with a
//     zero dy row by construction (margins of xp are zeroed so no NaNs).
// Traffic drops ~3-6x and MFMA work per staged byte grows 9x; the cost is
// (P+2)(Q+2)/(P*Q) extra (zero-row) reduce steps and the padded copy.
//
// Tile: 64(C) x 128(K); 8 waves as 2(C) x 4(K), wave patch 32x32;
// acc[9][2][2] f32x4 = 144 acc VGPRs -> 1 block/CU (2 waves/SIMD), LDS
// 4x8 KiB circular x + 2x16 KiB dy = 64 KiB.  Requires W <= 61 so the
// 9-seg read window [m'-(W+3), m'+63+(W+3)] spans <= 3 of the 4 slots.

constexpr int C9M = 64, C9N = 128;
// xp margin (bf16 elements worth of rows*C): staged blocks run one 64-row
// block past each end and fragment reads reach W+3 rows past that.
constexpr int C9_MARGIN_ROWS = 256;

DEVINL int c9x_byte(int row, int cbyte) {
  // [256 rows][64 ch] bf16 image, 128-B rows, chunk16 XOR swizzle
  return (row & 255) * 128 +
         ((((cbyte >> 4) ^ ((row & 3) << 1)) & 7) << 4 | (cbyte & 15));
}

// dy rows over the padded (P+2)x(Q+2) walk: zero at padding positions
struct DyPadRows {
  const bf16* dy;
  const bf16* zpage;
  int N, P, Q, K;
  // device state
  int n_[2], p_[2], q_[2];
  bool colok;
  int col, cgbyte;

  DEVINL void init(int col0, long m0, const int* myrow, int cg) {
    cgbyte = cg * 16;
    col = col0 + cg * 8;
    colok = col + 8 <= K;
    int PQ = (P + 2) * (Q + 2);
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      long m = m0 + myrow[p];
      n_[p] = (int)(m / PQ);
      int pq = (int)(m % PQ);
      p_[p] = pq / (Q + 2);
      q_[p] = pq % (Q + 2);
    }
  }
  DEVINL void step() {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      q_[i] += WBK;
      while (q_[i] >= Q + 2) {
        q_[i] -= Q + 2;
        p_[i] += 1;
      }
      while (p_[i] >= P + 2) {
        p_[i] -= P + 2;
        n_[i] += 1;
      }
    }
  }
  DEVINL const bf16* src(int p) const {
    const bf16* zp = (const bf16*)((const char*)zpage + cgbyte);
    if (n_[p] >= N || !colok) return zp;
    int pp = p_[p] - 1, qq = q_[p] - 1;
    if (pp < 0 || pp >= P || qq < 0 || qq >= Q) return zp;
    return dy + (((long)n_[p] * P + pp) * Q + qq) * K + col;
  }
};

__global__ __launch_bounds__(WT8, 1) void gemm_wg_c9_core(
    const bf16* __restrict__ xp, DyPadRows dyld, float* __restrict__ Cout,
    int C, int Ctot, int Kout, int W, long msteps_total, int nsplit,
    long strideSlot) {
  // xp: padded-copy base (margins zeroed); this block's channel tile c0
  __shared__ __align__(16) bf16 xbuf[4][64 * 64];   // circular, 4x8 KiB
  __shared__ __align__(16) bf16 dybuf[3][WBK * C9N];  // 3x16 KiB

  const int split = blockIdx.y;
  const long per = (msteps_total + nsplit - 1) / nsplit;
  const long k0 = split * per;
  const long ksteps = min(per, msteps_total - k0);
  if (ksteps <= 0) return;

  const int ntn = (Kout + C9N - 1) / C9N;
  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int c0 = (bid / ntn) * C9M, kn0 = (bid % ntn) * C9N;
  const int t = threadIdx.x;
  const int lw = t / WAVE, ll = t % WAVE;

  // ---- x staging: one contiguous 64-row (x 64-ch) block per step ----
  // 512 lanes x 16 B = 8 KiB: row = t/8, chunk = t%8, XOR-swizzled source
  const int xrow = t >> 3;
  const int xcg = (t & 7) ^ ((xrow & 3) << 1);
  // global source: xp + (m_abs + xrow)*Ctot + c0 + xcg*8
  const bf16* xsrc0 = xp + (long)(k0 * WBK + xrow) * Ctot + c0 + xcg * 8;
  const long xstep = (long)WBK * Ctot;

  auto stage_x = [&](long blk_rel) {  // block index relative to k0 (can be -1)
    unsigned slot = (unsigned)((k0 + blk_rel) & 3);
    unsigned lbase = __builtin_amdgcn_readfirstlane(
        (unsigned)(slot * 64 * 128 + lw * 1024));
    auto lds = (__attribute__((address_space(3))) unsigned int*)(
        (__attribute__((address_space(3))) char*)&xbuf[0][0] + lbase);
    auto g = (const __attribute__((address_space(1))) unsigned int*)(
        xsrc0 + blk_rel * xstep);
    __builtin_amdgcn_global_load_lds(g, lds, 16, 0, 0);
  };

  // ---- dy staging: gather rows, 2 glds passes (16 KiB) ----
  const int cg = (ll & 15) ^ ((ll >> 4) << 1);
  int myrow[2];
#pragma unroll
  for (int p = 0; p < 2; ++p) myrow[p] = lw * 8 + p * 4 + (ll >> 4);
  dyld.init(kn0, k0 * WBK, myrow, cg);

  auto stage_dy = [&](int buf) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      unsigned lbase =
          __builtin_amdgcn_readfirstlane((unsigned)((lw * 2 + p) * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&dybuf[buf][0] + lbase);
      auto g =
          (const __attribute__((address_space(1))) unsigned int*)dyld.src(p);
      __builtin_amdgcn_global_load_lds(g, lds, 16, 0, 0);
    }
    dyld.step();
  };

  f32x4 acc[9][2][2];
#pragma unroll
  for (int s = 0; s < 9; ++s)
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[s][i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // wave patch: 2(C) x 4(K) waves, each 32x32
  const int wm = (lw & 1) * 32, wn = (lw >> 1) * 32;
  const int ig = ll & 15, khi = ll >> 4;

  // seg row shifts in m': (r-1)*(W+2) + (s-1)
  int dseg[9];
#pragma unroll
  for (int r = 0; r < 3; ++r)
#pragma unroll
    for (int s = 0; s < 3; ++s) dseg[r * 3 + s] = (r - 1) * (W + 2) + (s - 1);

  auto xtrread = [&](int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)&xbuf[0][0] +
        c9x_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };
  auto dytrread = [&](const bf16* img, int row, int cbyte) {
    auto p = (__attribute__((address_space(3))) bf16x4*)(
        (__attribute__((address_space(3))) char*)img + tr_byte(row, cbyte));
    return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  };

  auto compute = [&](long ks) {
    int dbuf = (int)(ks % 3);
    int mbase = (int)((k0 + ks) & 3) * 64;  // circular base row of this step
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int rloc = kk * 32 + khi * 8 + (ig >> 2);
      int cb = (ig & 3) * 8;
      bf16x8 bfr[2];
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int cbyte = (wn + nf * 16) * 2 + cb;
        bf16x4 lo = dytrread(dybuf[dbuf], rloc, cbyte);
        bf16x4 hi = dytrread(dybuf[dbuf], rloc + 4, cbyte);
        bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
      // double-buffer the x fragments over segs: seg s+1's tr reads issue
      // before seg s's MFMAs so the ~50-cycle LDS latency hides under the
      // 4 MFMAs (the naive order measured 14% MFMA-busy / 23% issue-stall)
      bf16x8 af[2][2];
      auto lda = [&](int s, int buf) {
        int xr = mbase + rloc + dseg[s];
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
          int cbyte = (wm + mf * 16) * 2 + cb;
          bf16x4 lo = xtrread(xr, cbyte);
          bf16x4 hi = xtrread(xr + 4, cbyte);
          af[buf][mf] =
              __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
        }
      };
      lda(0, 0);
#pragma unroll
      for (int s = 0; s < 9; ++s) {
        if (s + 1 < 9) lda(s + 1, (s + 1) & 1);
#pragma unroll
        for (int mf = 0; mf < 2; ++mf)
#pragma unroll
          for (int nf = 0; nf < 2; ++nf)
            acc[s][mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[s & 1][mf], bfr[nf], acc[s][mf][nf], 0, 0, 0);
      }
    }
  };

  // prologue: x blocks -1, 0, 1 (margins make -1 and past-end valid), dy 0
  stage_x(-1);
  stage_x(0);
  stage_x(1);
  stage_dy(0);
  if (ksteps > 1) stage_dy(1);
  __syncthreads();
  for (long ks = 0; ks < ksteps; ++ks) {
    stage_x(ks + 2);
    if (ks + 2 < ksteps) stage_dy((int)((ks + 2) % 3));
    compute(ks);
    __syncthreads();
  }

  // epilogue: 9 seg tiles into this split's slot [seg][C][K]
  float* cslot = Cout + (long)split * strideSlot;
#pragma unroll
  for (int s = 0; s < 9; ++s) {
    float* cseg = cslot + (long)s * C * Kout;
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int gcol = kn0 + wn + nf * 16 + (ll & 15);
        if (gcol >= Kout) continue;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int grow = c0 + wm + mf * 16 + (ll >> 4) * 4 + i;
          if (grow >= C) continue;
          cseg[(long)grow * Kout + gcol] = acc[s][mf][nf][i];
        }
      }
    }
  }
}

// padded copy: xp[(n, hp, wp, c)] over (H+2)x(W+2) with 1-px zero ring;
// front/back margins of C9_MARGIN_ROWS*C elements are zeroed separately.
__global__ __launch_bounds__(256) void pad_image_kernel(
    const bf16* __restrict__ x, bf16* __restrict__ xp, int N, int H, int W,
    int C, long total8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total8) return;
  long e = i * 8;  // 8 channels per thread; C % 8 == 0
  int c = (int)(e % C);
  long pix = e / C;
  int wp = (int)(pix % (W + 2));
  long t2 = pix / (W + 2);
  int hp = (int)(t2 % (H + 2));
  int n = (int)(t2 / (H + 2));
  bf16x8 v = {};
  if (hp >= 1 && hp <= H && wp >= 1 && wp <= W)
    v = *(const bf16x8*)(x + (((long)n * H + hp - 1) * W + wp - 1) * C + c);
  *(bf16x8*)(xp + e) = v;
}

__global__ __launch_bounds__(256) void zero_bf16_kernel(bf16* p, long n8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n8) ((bf16x8*)p)[i] = bf16x8{};
}

bool conv_wg_c9_eligible(int R, int S, int sh, int sw, int ph, int pw,
                         int dh, int dw, int H, int W, int P, int Q, int C,
                         int K) {
  // W >= 24: the padded walk costs (H+2)(W+2)/(HW) extra reduce steps —
  // measured (profiles, r2 A/B at b256): +14% at 56x56, tie at 28x28,
  // -15% at 14x14 (1.31x pad overhead) — so small images keep the
  // per-seg path.
  return R == 3 && S == 3 && sh == 1 && sw == 1 && ph == 1 && pw == 1 &&
         dh == 1 && dw == 1 && P == H && Q == W && W <= 61 && W >= 24 &&
         C % 8 == 0 && K % 8 == 0;
}

long conv_wg_c9_xp_elems(int N, int H, int W, int C) {
  return ((long)N * (H + 2) * (W + 2) + 2L * C9_MARGIN_ROWS) * C;
}

// ---------------------------------------------------------------------------
// Split-slot fold epilogue
// ---------------------------------------------------------------------------
// Split-K blocks write per-split f32 slot buffers with plain stores (no
// HBM atomics, no pre-zero memset — the round-1 PMC evidence showed the
// small-tile wgrads wait-bound on exactly those).  This single pass then
// sums the slots in f32 (deterministic, fixed order) and emits the final
// bf16 tensor, fusing the f32->bf16 cast and, in the TRANSPOSE variant,
// the [RSC,K] -> [K,RSC] weight-grad permute that used to be separate
// torch launches after every wgrad.

// At high split counts the final tensor is small (one or a few tiles), so
// a one-stage fold would be a tiny latency-bound grid doing up to 128
// serial slot loads per element.  Two levels instead: a WIDE partial stage
// (grid.y = slot groups of 16) halves into f32 partial slots appended
// after the input slots, then the final fold reads <= 8 of those.
constexpr int WG_FOLD_GROUP = 16;

__global__ __launch_bounds__(256) void wg_fold_partial(
    const float* __restrict__ work, long slotStride, int nslots, long n,
    float* __restrict__ outslots) {
  long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (base >= n) return;
  int cnt = min(WG_FOLD_GROUP, nslots - (int)blockIdx.y * WG_FOLD_GROUP);
  const float* p =
      work + (long)blockIdx.y * WG_FOLD_GROUP * slotStride + base;
  // 4 independent accumulators keep >=4 loads in flight per lane — the
  // 2-deep version measured latency-bound (95% wave-parked)
  f32x4 s0{0.f, 0.f, 0.f, 0.f}, s1{0.f, 0.f, 0.f, 0.f};
  f32x4 s2{0.f, 0.f, 0.f, 0.f}, s3{0.f, 0.f, 0.f, 0.f};
  int sl = 0;
  for (; sl + 4 <= cnt; sl += 4) {
    s0 += *(const f32x4*)p;
    s1 += *(const f32x4*)(p + slotStride);
    s2 += *(const f32x4*)(p + 2 * slotStride);
    s3 += *(const f32x4*)(p + 3 * slotStride);
    p += 4 * slotStride;
  }
  for (; sl < cnt; ++sl) {
    s0 += *(const f32x4*)p;
    p += slotStride;
  }
  s0 += s1 + s2 + s3;
  *(f32x4*)(outslots + (long)blockIdx.y * slotStride + base) = s0;
}

__global__ __launch_bounds__(256) void wg_fold_flat(
    const float* __restrict__ work, long slotStride, int nslots, long n,
    bf16* __restrict__ out) {
  long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (base >= n) return;
  f32x4 s0{0.f, 0.f, 0.f, 0.f}, s1{0.f, 0.f, 0.f, 0.f};
  f32x4 s2{0.f, 0.f, 0.f, 0.f}, s3{0.f, 0.f, 0.f, 0.f};
  const float* p = work + base;
  int sl = 0;
  for (; sl + 4 <= nslots; sl += 4) {
    s0 += *(const f32x4*)p;
    s1 += *(const f32x4*)(p + slotStride);
    s2 += *(const f32x4*)(p + 2 * slotStride);
    s3 += *(const f32x4*)(p + 3 * slotStride);
    p += 4 * slotStride;
  }
  for (; sl < nslots; ++sl) {
    s0 += *(const f32x4*)p;
    p += slotStride;
  }
  s0 += s1 + s2 + s3;
  bf16x4 o{f2bf(s0.x), f2bf(s0.y), f2bf(s0.z), f2bf(s0.w)};
  *(bf16x4*)(out + base) = o;
}

__global__ __launch_bounds__(256) void wg_fold_tr(
    const float* __restrict__ work, long slotStride, int nslots, long rows,
    long cols, bf16* __restrict__ out) {
  __shared__ float tile[32][33];  // +1: no bank conflicts on the transpose
  long ntr = (rows + 31) / 32;
  long tr = (long)(blockIdx.x % ntr) * 32;
  long tc = (long)(blockIdx.x / ntr) * 32;
  int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;  // 32 x 8
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    int r = ty + p * 8;
    long gr = tr + r, gc = tc + tx;
    float s0 = 0.f, s1 = 0.f;
    if (gr < rows && gc < cols) {
      const float* q = work + gr * cols + gc;
      int sl = 0;
      for (; sl + 2 <= nslots; sl += 2) {
        s0 += q[0];
        s1 += q[slotStride];
        q += 2 * slotStride;
      }
      if (sl < nslots) s0 += q[0];
    }
    tile[r][tx] = s0 + s1;
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    int c = ty + p * 8;
    long gc = tc + c, gr = tr + tx;
    if (gc < cols && gr < rows) out[gc * rows + gr] = f2bf(tile[tx][c]);
  }
}

// slots the CALLER must allocate for `used` split slots (adds headroom
// for the partial stage)
static int wg_alloc_slots(int used) {
  return used + (used > WG_FOLD_GROUP
                     ? (used + WG_FOLD_GROUP - 1) / WG_FOLD_GROUP
                     : 0);
}

void wg_fold(const float* work, long slotStride, int nslots, long rows,
             long cols, bf16* out, bool transpose, hipStream_t stream) {
  long n = rows * cols;  // cols % 8 == 0 -> n % 4 == 0
  const float* src = work;
  if (nslots > WG_FOLD_GROUP) {
    int nchunk = (nslots + WG_FOLD_GROUP - 1) / WG_FOLD_GROUP;
    float* partial = const_cast<float*>(work) + (long)nslots * slotStride;
    long blocks = (n / 4 + 255) / 256;
    wg_fold_partial<<<dim3((unsigned)blocks, nchunk), 256, 0, stream>>>(
        work, slotStride, nslots, n, partial);
    src = partial;
    nslots = nchunk;
  }
  if (transpose) {
    long ntr = (rows + 31) / 32, ntc = (cols + 31) / 32;
    wg_fold_tr<<<dim3((unsigned)(ntr * ntc)), 256, 0, stream>>>(
        src, slotStride, nslots, rows, cols, out);
  } else {
    long blocks = (n / 4 + 255) / 256;
    wg_fold_flat<<<dim3((unsigned)blocks), 256, 0, stream>>>(
        src, slotStride, nslots, n, out);
  }
}

static bool wg_use_zip() {
  static const bool z = [] {
    const char* e = getenv("MXA_WG_ZIP");
    return e && e[0] == '1';
  }();
  return z;
}

static int wg_ring() {
  // 0 (default) = 2-phase tr kernel; 4 = 64-KiB ring (2 blocks/CU);
  // 8 = tn8-depth ring (128 KiB, 1 block/CU).  Measured on MI355X
  // (profiles/ring{0,4,8}_wg.log): the ring LOSES ~10% (RING=4) to ~2x
  // (RING=8) on every wgrad shape — the 2-phase schedule already overlaps
  // one full tile of DMA with compute, and halving the barrier quantum to
  // 32 rows doubles the s_barrier rate for only 8 MFMA per wave between
  // barriers.  The small-tile wgrad wait time is split-K epilogue atomics
  // and launch overhead, not in-loop DMA depth.  Kept as a measured
  // negative result and for future schedule experiments.
  static const int r = [] {
    const char* e = getenv("MXA_WG_RING");
    return e ? atoi(e) : 0;
  }();
  return (r == 4 || r == 8) ? r : 0;
}

template <bool A_IM2COL, bool OUTBF16>
static void launch_wg_ring(const TrPlainRows1& a, const TrIm2colRows1& ai,
                           const TrPlainRows1& b, float* C, int Mo, int No,
                           long ldc, long strideSeg, int S, int dh, int dw,
                           int hsteps, int nsplit, int use_atomic, ZMap za,
                           ZMap zb, ZMap zc, dim3 grid, hipStream_t stream) {
  if (wg_ring() == 8)
    gemm_wg_ring_core<A_IM2COL, OUTBF16, 8><<<grid, WT8, 0, stream>>>(
        a, ai, b, C, Mo, No, ldc, strideSeg, S, dh, dw, hsteps, nsplit,
        use_atomic, za, zb, zc);
  else
    gemm_wg_ring_core<A_IM2COL, OUTBF16, 4><<<grid, WT8, 0, stream>>>(
        a, ai, b, C, Mo, No, ldc, strideSeg, S, dh, dw, hsteps, nsplit,
        use_atomic, za, zb, zc);
}

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------

static int wg_split_target() {
  static const int t = [] {
    const char* e = getenv("MXA_WG_TARGET");
    return e ? atoi(e) : 1024;
  }();
  return t;
}

static int wg_min_ksteps() {
  // 48 since the slot-fold epilogue: fold traffic scales with the split
  // count, so deeper splits-per-slot win (b512 sweep: mk=48 8169-8199 vs
  // mk=24 8075-8170 samples/s across fill targets)
  static const int t = [] {
    const char* e = getenv("MXA_WG_MINK");
    return e ? atoi(e) : 48;
  }();
  return t;
}

static int pick_split(long blocks_xy, long ksteps) {
  // Fill the chip (~512 blocks of 4 waves at 2 waves/SIMD), but every
  // split adds a 64-KiB f32 atomic finish per block — so each split must
  // also amortize over enough reduce depth (>= MXA_WG_MINK k-steps,
  // default 24).  Measured on MI355X: dropping the old fill-only target
  // lifted small-Mo wgrads 107->190 TF (atomic-bound), while res2-3x3
  // (1 tile) still needs ~128 splits for occupancy.
  int goal = wg_split_target();
  int mink = wg_min_ksteps();
  long depth_cap = mink > 0 ? ksteps / mink : ksteps;
  int target = 1;
  while (blocks_xy * target < goal && target < ksteps && target < 128 &&
         (blocks_xy * target < 256 || target < depth_cap))
    target *= 2;
  return target;
}

struct WgPlanT {
  int nsplit;  // grid.y factor
  int used;    // splits that actually produce output (slots to fold)
  int ksteps;  // 64-row reduce steps
  int hsteps;  // 32-row reduce steps (ring kernel)
};

static WgPlanT wg_plan_(long bxy, long mtotal) {
  WgPlanT p;
  p.ksteps = (int)((mtotal + WBK - 1) / WBK);
  p.hsteps = (int)((mtotal + 31) / 32);
  p.nsplit = pick_split(bxy, p.ksteps);
  int steps = wg_ring() ? p.hsteps : p.ksteps;
  int per = (steps + p.nsplit - 1) / p.nsplit;
  p.used = (steps + per - 1) / per;
  return p;
}

static bool wg_w64() {
  static const bool v = [] {
    const char* e = getenv("MXA_WG64");
    return e && e[0] == '1';  // off until A/B'd
  }();
  return v;
}

int gemm_wg_nslots(long mtotal, int Mo, int No, bool transpose) {
  // 0 = direct bf16 output, no workspace (single split, plain layout,
  // kernels that have the OUTBF16 epilogue)
  WgPlanT p = wg_plan_((long)cdiv_h(Mo, WBM) * cdiv_h(No, WBN), mtotal);
  bool tr_path = !wg_use_zip() && Mo % 8 == 0 && No % 8 == 0;
  if (wg_w64() && Mo >= 256 && tr_path) {
    // the 256x128 w64 plan may pick a different split count; size the
    // workspace for whichever path will actually run
    WgPlanT p64 = wg_plan_((long)cdiv_h(Mo, 256) * cdiv_h(No, 128),
                           mtotal);
    if (p64.used == 1 && !transpose) return 0;
    return wg_alloc_slots(p64.used);
  }
  if (p.used == 1 && !transpose && tr_path) return 0;
  return wg_alloc_slots(p.used);
}

void colsum_bf16(const bf16* x, float* out, long m, int N, hipStream_t s);

void gemm_wg_bf16(const bf16* A, const bf16* B, float* work, bf16* out,
                  bool transpose, long mtotal, int Mo, int No, long lda,
                  long ldb, const bf16* zpage, hipStream_t stream,
                  float* biasg) {
  long bxy = (long)cdiv_h(Mo, WBM) * cdiv_h(No, WBN);
  WgPlanT pl = wg_plan_(bxy, mtotal);
  dim3 grid((unsigned)bxy, pl.nsplit, 1);
  long slotStride = (long)Mo * No;
  const bool direct = work == nullptr;  // OUTBF16 epilogue, no fold pass
  float* Cdst = direct ? (float*)out : work;
  // bias-grad fusion lives in the tr core; other paths fall back to the
  // standalone colsum over A (same result, one extra read of A)
  bool bias_fused = false;
  if (!wg_use_zip() && lda % 8 == 0 && ldb % 8 == 0) {
    if (wg_ring()) {
      TrPlainRows1 a{A, zpage, mtotal, lda, Mo};
      TrPlainRows1 b{B, zpage, mtotal, ldb, No};
      TrIm2colRows1 dummy{};
      if (direct)
        launch_wg_ring<false, true>(a, dummy, b, Cdst, Mo, No, No, 0, 1, 1,
                                    1, pl.hsteps, pl.nsplit, 0, ZMap{1, 0, 0},
                                    ZMap{1, 0, 0}, ZMap{1, 0, 0}, grid,
                                    stream);
      else
        gemm_wg_ring_core<false, false, 4><<<grid, WT8, 0, stream>>>(
            a, dummy, b, Cdst, Mo, No, No, 0, 1, 1, 1, pl.hsteps, pl.nsplit,
            0, ZMap{1, 0, 0}, ZMap{1, 0, 0}, ZMap{1, 0, 0}, slotStride);
    } else {
      if (wg_w64() && Mo >= 256) {
        long bxy64 = (long)cdiv_h(Mo, 256) * cdiv_h(No, 128);
        WgPlanT p64 = wg_plan_(bxy64, mtotal);
        dim3 g64((unsigned)bxy64, p64.nsplit, 1);
        TrPlainRows a{A, zpage, mtotal, lda, Mo};
        TrPlainRows b{B, zpage, mtotal, ldb, No};
        if (direct)
          gemm_wg_w64_core<true><<<g64, WT8, 0, stream>>>(
              a, b, Cdst, Mo, No, No, p64.ksteps, p64.nsplit, 0, biasg);
        else
          gemm_wg_w64_core<false><<<g64, WT8, 0, stream>>>(
              a, b, Cdst, Mo, No, No, p64.ksteps, p64.nsplit, slotStride,
              biasg);
        if (!direct)
          wg_fold(work, slotStride, p64.used, Mo, No, out, transpose,
                  stream);
        if (biasg) { /* fused above */ }
        return;
      }
      TrPlainRows a{A, zpage, mtotal, lda, Mo};
      TrPlainRows b{B, zpage, mtotal, ldb, No};
      TrIm2colRows dummy{};
      if (direct)
        gemm_wg_tr_core<false, true><<<grid, WT8, 0, stream>>>(
            a, dummy, b, Cdst, Mo, No, No, 0, 1, 1, 1, pl.ksteps, pl.nsplit,
            0, ZMap{1, 0, 0}, ZMap{1, 0, 0}, ZMap{1, 0, 0}, 0, biasg);
      else
        gemm_wg_tr_core<false, false><<<grid, WT8, 0, stream>>>(
            a, dummy, b, Cdst, Mo, No, No, 0, 1, 1, 1, pl.ksteps, pl.nsplit,
            0, ZMap{1, 0, 0}, ZMap{1, 0, 0}, ZMap{1, 0, 0}, slotStride,
            biasg);
      bias_fused = true;
    }
  } else {
    WgPlainRows a{A, zpage, mtotal, lda, Mo};
    WgPlainRows b{B, zpage, mtotal, ldb, No};
    WgIm2colRows dummy{};
    gemm_wg_core<false><<<grid, WTHREADS, 0, stream>>>(
        a, dummy, b, Cdst, Mo, No, No, 0, 1, 1, 1, pl.ksteps, pl.nsplit, 0,
        slotStride);
  }
  if (!direct)
    wg_fold(work, slotStride, pl.used, Mo, No, out, transpose, stream);
  if (biasg && !bias_fused) colsum_bf16(A, biasg, mtotal, Mo, stream);
}

void conv_wg_pad_image(const bf16* x, bf16* xp_alloc, int N, int H, int W,
                       int C, hipStream_t stream) {
  long margin8 = (long)C9_MARGIN_ROWS * C / 8;
  long body = (long)N * (H + 2) * (W + 2) * C;
  zero_bf16_kernel<<<dim3((unsigned)((margin8 + 255) / 256)), 256, 0,
                     stream>>>(xp_alloc, margin8);
  zero_bf16_kernel<<<dim3((unsigned)((margin8 + 255) / 256)), 256, 0,
                     stream>>>(xp_alloc + (long)C9_MARGIN_ROWS * C + body,
                               margin8);
  long total8 = body / 8;
  pad_image_kernel<<<dim3((unsigned)((total8 + 255) / 256)), 256, 0,
                     stream>>>(x, xp_alloc + (long)C9_MARGIN_ROWS * C, N, H,
                               W, C, total8);
}

static WgPlanT c9_plan(int N, int H, int W, int C, int Kout) {
  long m2 = (long)N * (H + 2) * (W + 2);
  long bxy = (long)cdiv_h(C, C9M) * cdiv_h(Kout, C9N);
  WgPlanT p;
  p.ksteps = (int)((m2 + WBK - 1) / WBK);
  p.hsteps = p.ksteps;
  p.nsplit = pick_split(bxy, p.ksteps);
  int per = (p.ksteps + p.nsplit - 1) / p.nsplit;
  p.used = (p.ksteps + per - 1) / per;
  return p;
}

int conv2d_wgrad_c9_nslots(int N, int H, int W, int C, int Kout) {
  return wg_alloc_slots(c9_plan(N, H, W, C, Kout).used);
}

void conv2d_wgrad_c9_bf16(const bf16* xp_alloc, const bf16* dy, float* work,
                          bf16* out_krsc, int N, int H, int W, int C,
                          int Kout, const bf16* zpage, hipStream_t stream) {
  WgPlanT pl = c9_plan(N, H, W, C, Kout);
  long bxy = (long)cdiv_h(C, C9M) * cdiv_h(Kout, C9N);
  dim3 grid((unsigned)bxy, pl.nsplit, 1);
  DyPadRows dyld{dy, zpage, N, H, W, Kout};
  long strideSlot = 9L * C * Kout;
  gemm_wg_c9_core<<<grid, WT8, 0, stream>>>(
      xp_alloc + (long)C9_MARGIN_ROWS * C, dyld, work, C, C, Kout, W,
      pl.ksteps, pl.nsplit, strideSlot);
  wg_fold(work, strideSlot, pl.used, 9L * C, Kout, out_krsc, true, stream);
}

int conv2d_wgrad_nslots(long mtotal, int C, int Kout, int R, int S) {
  WgPlanT p =
      wg_plan_((long)cdiv_h(C, WBM) * cdiv_h(Kout, WBN) * R * S, mtotal);
  return wg_alloc_slots(p.used);
}

void conv2d_wgrad_bf16(const bf16* x, const bf16* dy, float* work,
                       bf16* out_krsc, int Nb, int H, int W, int C, int Kout,
                       int R, int S, int P, int Q, int sh, int sw, int ph,
                       int pw, int dh, int dw_, const bf16* zpage,
                       hipStream_t stream) {
  long mtotal = (long)Nb * P * Q;
  long bxy = (long)cdiv_h(C, WBM) * cdiv_h(Kout, WBN);
  WgPlanT pl = wg_plan_(bxy * R * S, mtotal);
  dim3 grid((unsigned)bxy, R * S * pl.nsplit, 1);
  long strideSeg = (long)C * Kout;          // within a slot: [R*S][C][K]
  long slotStride = strideSeg * R * S;      // between slots
  if (!wg_use_zip() && C % 8 == 0 && Kout % 8 == 0) {
    if (wg_ring()) {
      TrIm2colRows1 a{x, zpage, Nb, H, W, C, P, Q, sh, sw, ph, pw};
      TrPlainRows1 b{dy, zpage, mtotal, (long)Kout, Kout};
      TrPlainRows1 dummy{};
      gemm_wg_ring_core<true, false, 4><<<grid, WT8, 0, stream>>>(
          dummy, a, b, work, C, Kout, Kout, strideSeg, S, dh, dw_, pl.hsteps,
          pl.nsplit, 0, ZMap{1, 0, 0}, ZMap{1, 0, 0}, ZMap{1, 0, 0},
          slotStride);
    } else {
      TrIm2colRows a{x, zpage, Nb, H, W, C, P, Q, sh, sw, ph, pw};
      TrPlainRows b{dy, zpage, mtotal, (long)Kout, Kout};
      TrPlainRows dummy{};
      gemm_wg_tr_core<true><<<grid, WT8, 0, stream>>>(
          dummy, a, b, work, C, Kout, Kout, strideSeg, S, dh, dw_, pl.ksteps,
          pl.nsplit, 0, ZMap{1, 0, 0}, ZMap{1, 0, 0}, ZMap{1, 0, 0},
          slotStride);
    }
  } else {
    WgIm2colRows a{x, zpage, Nb, H, W, C, P, Q, sh, sw, ph, pw};
    WgPlainRows b{dy, zpage, mtotal, (long)Kout, Kout};
    WgPlainRows dummy{};
    gemm_wg_core<true><<<grid, WTHREADS, 0, stream>>>(
        dummy, a, b, work, C, Kout, Kout, strideSeg, S, dh, dw_, pl.ksteps,
        pl.nsplit, 0, slotStride);
  }
  // fold all slots, transposing [R*S*C][K] -> [K][R*S*C] == (K,R,S,C)
  wg_fold(work, slotStride, pl.used, (long)R * S * C, Kout, out_krsc, true,
          stream);
}

// Batched attention dK/dV over per-(batch, head) slices (z = b*H + h):
//   dK[z][tk][d] = sum_tq dS[z][tq][tk] * Q[b, tq, h*D + d]   (and
//   dV with dS := P, Q := dO).  A is the contiguous [BH, Tq, X] probs /
//   score-grads tensor; B and the f32 output live in [B, S, H*D] layout
//   read/written in place.
void attn_wg_bf16(const bf16* A, const bf16* Bq, bf16* C, int B, int H,
                  int Tq, int Tk, int D, const bf16* zpage,
                  hipStream_t stream) {
  long hd = (long)H * D;
  int ksteps = cdiv_h(Tq, WBK);
  long bxy = (long)cdiv_h(Tk, WBM) * cdiv_h(D, WBN);
  dim3 grid((unsigned)bxy, 1, (unsigned)(B * H));
  // single reduce slice per z -> no atomics -> bf16 output directly
  // (saves a full f32->bf16 cast pass over [B, S, H*D] per call)
  if (wg_ring()) {
    TrPlainRows1 a{A, zpage, Tq, (long)Tk, Tk};
    TrPlainRows1 b{Bq, zpage, Tq, hd, D};
    TrIm2colRows1 dummy{};
    launch_wg_ring<false, true>(a, dummy, b, (float*)C, Tk, D, hd, 0, 1, 1,
                                1, cdiv_h(Tq, 32), 1, 0,
                                ZMap{1, (long)Tq * Tk, 0}, ZMap{H, Tq * hd, D},
                                ZMap{H, Tk * hd, D}, grid, stream);
    return;
  }
  TrPlainRows a{A, zpage, Tq, (long)Tk, Tk};
  // rowlen D (not H*D): the per-head slice is only D columns wide — a
  // wider guard would let edge lanes read past the allocation on the
  // final row of the final batch.
  TrPlainRows b{Bq, zpage, Tq, hd, D};
  TrIm2colRows dummy{};
  gemm_wg_tr_core<false, true><<<grid, WT8, 0, stream>>>(
      a, dummy, b, (float*)C, Tk, D, hd, 0, 1, 1, 1, ksteps, 1, 0,
      ZMap{1, (long)Tq * Tk, 0}, ZMap{H, Tq * hd, D}, ZMap{H, Tk * hd, D});
}

}  // namespace mxa
