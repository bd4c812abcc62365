// Fused flash-attention for gfx950 (MI355X), D = 64, any Tq/Tk
// (tail blocks clamp their staging rows and mask scores at col >= Tk;
// cross-attention Tq != Tk supported for the non-causal case).
//
// Forward: one workgroup per (batch*head, 128-query block); K/V stream
// through double-buffered LDS in 128-row chunks; scores, online softmax,
// dropout and the P@V accumulation all stay on-chip — S/P/mask are never
// materialized in HBM (reference parity: contrib/transformer.cu composed
// GEMMs + softmax; here the MI355X-native fused form).  Saves per-row
// softmax stats (m, l) for the exact backward.
//
// Layouts (empirically validated building blocks from gemm_tn/gemm_nn):
//   Q/K images  [128 rows][64 k]  — TN style, glds + T2 XOR swizzle,
//                                   ds_read_b128 fragments
//   V image     [128 tk][64 d]    — row-major + (row&3)<<1 chunk swizzle,
//                                   ds_read_b64_tr_b16 fragments
//   P image     [128 q][128 tk]   — TN style with 16-key swizzle
//
// Wave layout (512 threads): 8 waves as 4(M=q) x 2(N); each wave owns the
// SAME 32 query rows in the S phase (N = tk half) and the O phase
// (N = d half), so the online-softmax rescale of the O accumulator is a
// per-lane multiply.  Cross-wave row stats go through a 2-KiB LDS
// scratch (the two tk-half waves of each row pair).
#include "common.h"

namespace mxa {

DEVINL int fa_qk_off(int row, int chunk16) {
  // [128][64 bf16] image, 128-B rows (T2 swizzle, 8 chunks/row)
  return row * 64 + ((chunk16 ^ (row & 7)) << 3);
}

DEVINL int fa_p_byte(int row, int chunk16) {
  // [128][128 bf16] image, 256-B rows, 16 chunks/row, 16-key swizzle
  return row * 256 + ((chunk16 ^ (row & 15)) << 4);
}

DEVINL int fa_v_byte(int row, int cbyte) {
  // [128][64 bf16] row-major tr image (see gemm_nn nn_b64_byte)
  return row * 128 + ((((cbyte >> 4) ^ ((row & 3) << 1)) << 4) | (cbyte & 15));
}

// raw barrier preceded by lgkmcnt(0): publishes every wave's LDS writes
// without the vmcnt(0) drain __syncthreads folds in while a glds is in
// flight — that drain serialized the kv prefetch against each softmax
// reduction barrier (PMC: 45% wave-park on flash_fwd_core)
#define FA_LDS_BARRIER()                                   \
  do {                                                     \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");     \
    __builtin_amdgcn_s_barrier();                          \
  } while (0)

DEVINL int fa_tr128_byte(int row, int cbyte) {
  // [128][128 bf16] row-major tr image (256-B rows), same chunk key
  return row * 256 + ((((cbyte >> 4) ^ ((row & 3) << 1)) << 4) | (cbyte & 15));
}

DEVINL int fa_qk128_off(int row, int chunk16) {
  // element offset in a [*][128 bf16] TN image (256-B rows, 16 chunks)
  return row * 128 + ((chunk16 ^ (row & 15)) << 3);
}

// ---------------------------------------------------------------------------
// D = 128 forward: same structure, KV streamed in 64-row chunks so the
// wider head fits LDS (q 32K + k/v 2x16K each + p 16K = 114 KiB).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(512, 1) void flash_fwd_core_d128(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O,
    float* __restrict__ Ms, float* __restrict__ Ls, int B, int H, int Tq,
    int Tk, float alpha, int causal, const u64* __restrict__ seed_ptr,
    u64 salt, float pdrop, float dscale) {
  constexpr int D = 128;
  __shared__ __align__(16) bf16 qimg[128 * D];
  __shared__ __align__(16) bf16 kimg[2][64 * D];
  __shared__ __align__(16) bf16 vimg[2][64 * D];
  __shared__ __align__(16) bf16 pimg[128 * 64];
  __shared__ float sred[2][128][2];

  const int z = blockIdx.z;
  const int b = z / H, h = z % H;
  const int qb = blockIdx.x;
  const long hd = (long)H * D;
  const int t = threadIdx.x;
  const int lw = t / WAVE, ll = t % WAVE;

  const u64 seed = seed_ptr ? seed_finalize(*seed_ptr + salt) : 0;
  const u32 thresh = (u32)(pdrop * 4294967296.0f);

  // staging geometry for 256-B rows: one glds pass = 32 rows (wave = 4
  // rows); TN images swizzle on 16 chunks with key row&15, the V tr
  // image with key (row&3)<<1
  const int srow4 = lw * 4 + (ll >> 4);  // row within a 32-row pass
  const int qk_cg = (ll & 15) ^ (srow4 & 15);
  const int v_cg = (ll & 15) ^ ((srow4 & 3) << 1);

  {
    const bf16* qbase = Q + (long)b * Tq * hd + (long)h * D;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      int row = p * 32 + srow4;
      unsigned lbase = __builtin_amdgcn_readfirstlane(
          (unsigned)((p * 32 + lw * 4) * 256));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&qimg[0] + lbase);
      int qrow = min(qb * 128 + row, Tq - 1);
      const bf16* gp = qbase + (long)qrow * hd + qk_cg * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds,
          16, 0, 0);
    }
  }

  const bf16* kbase = K + (long)b * Tk * hd + (long)h * D;
  const bf16* vbase = V + (long)b * Tk * hd + (long)h * D;
  auto stage_kv = [&](int buf, int c) {  // 64 kv rows
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      int row = p * 32 + srow4;
      unsigned lbase = __builtin_amdgcn_readfirstlane(
          (unsigned)((p * 32 + lw * 4) * 256));
      auto ldsK = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&kimg[buf][0] + lbase);
      auto ldsV = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&vimg[buf][0] + lbase);
      int krow = min(c * 64 + row, Tk - 1);
      const bf16* gk = kbase + (long)krow * hd + qk_cg * 8;
      const bf16* gv = vbase + (long)krow * hd + v_cg * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gk, ldsK,
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gv, ldsV,
          16, 0, 0);
    }
  };

  const int wmq = (lw & 3) * 32;   // this wave's 32 query rows
  const int wnt = (lw >> 2) * 32;  // S phase: tk half (of 64)
  const int wno = (lw >> 2) * 64;  // O phase: d half (of 128)
  const int lrow = ll & 15, lk = ll >> 4;

  float m_i[2][4], l_i[2][4];
  f32x4 acc_o[2][4];
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      m_i[mf][i] = -3.0e38f;
      l_i[mf][i] = 0.f;
    }
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) acc_o[mf][nf] = f32x4{0, 0, 0, 0};

  const int nchunks_all = (Tk + 63) / 64;
  const int nchunks = causal ? min(2 * qb + 2, nchunks_all) : nchunks_all;

  stage_kv(0, 0);
  __syncthreads();

  for (int c = 0; c < nchunks; ++c) {
    int buf = c & 1;
    if (c + 1 < nchunks) stage_kv(buf ^ 1, c + 1);

    // ---- S = alpha * Q K^T (wave: 32 q x 32 tk), reduce d = 128 ----
    f32x4 acc_s[2][2];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) acc_s[mf][nf] = f32x4{0, 0, 0, 0};
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      bf16x8 af[2], bfr[2];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        af[mf] = *(const bf16x8*)&qimg[fa_qk128_off(
            wmq + mf * 16 + lrow, chunk)];
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        bfr[nf] = *(const bf16x8*)&kimg[buf][fa_qk128_off(
            wnt + nf * 16 + lrow, chunk)];
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc_s[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc_s[mf][nf], 0, 0, 0);
    }

    // ---- scale + masks + chunk row-max ----
    float cmax[2][4];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) cmax[mf][i] = -3.0e38f;
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int col = c * 64 + wnt + nf * 16 + (ll & 15);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int row = qb * 128 + wmq + mf * 16 + (ll >> 4) * 4 + i;
          float v = acc_s[mf][nf][i] * alpha;
          if ((causal && col > row) || col >= Tk) v = -3.0e38f;
          acc_s[mf][nf][i] = v;
          cmax[mf][i] = fmaxf(cmax[mf][i], v);
        }
      }
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float v = cmax[mf][i];
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          v = fmaxf(v, __shfl_xor(v, off, 64));
        cmax[mf][i] = v;
      }
    if ((ll & 15) == 0) {
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int i = 0; i < 4; ++i)
          sred[0][wmq + mf * 16 + (ll >> 4) * 4 + i][lw >> 2] =
              cmax[mf][i];
    }
    FA_LDS_BARRIER();
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int r = wmq + mf * 16 + (ll >> 4) * 4 + i;
        cmax[mf][i] = fmaxf(sred[0][r][0], sred[0][r][1]);
      }

    // ---- online rescale + P = exp(s - m_new) + chunk row-sum ----
    float csum[2][4];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float m_new = fmaxf(m_i[mf][i], cmax[mf][i]);
        float f = __expf(m_i[mf][i] - m_new);
        m_i[mf][i] = m_new;
        l_i[mf][i] *= f;
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) acc_o[mf][nf][i] *= f;
        csum[mf][i] = 0.f;
      }
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          float p = __expf(acc_s[mf][nf][i] - m_i[mf][i]);
          acc_s[mf][nf][i] = p;
          csum[mf][i] += p;
        }
      }
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float v = csum[mf][i];
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          v += __shfl_xor(v, off, 64);
        csum[mf][i] = v;
      }
    if ((ll & 15) == 0) {
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int i = 0; i < 4; ++i)
          sred[1][wmq + mf * 16 + (ll >> 4) * 4 + i][lw >> 2] =
              csum[mf][i];
    }
    FA_LDS_BARRIER();
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int r = wmq + mf * 16 + (ll >> 4) * 4 + i;
        l_i[mf][i] += sred[1][r][0] + sred[1][r][1];
      }

    // ---- dropout + write P image [128 q][64 tk] ----
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int col = wnt + nf * 16 + (ll & 15);
        int gcol = c * 64 + col;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int row = wmq + mf * 16 + (ll >> 4) * 4 + i;
          float p = acc_s[mf][nf][i];
          if (thresh) {
            long grow = (long)qb * 128 + row;
            u32 r = hash_u32(seed, ((long)z * Tq + grow) * Tk + gcol);
            p = r >= thresh ? p * dscale : 0.f;
          }
          *(bf16*)((char*)pimg + fa_qk_off(row, col >> 3) * 2 +
                   (col & 7) * 2) = f2bf(p);
        }
      }
    __syncthreads();

    // ---- acc_o += P~ V (wave: 32 q x 64 d), K dim = 64 tk ----
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[2], bfr[4];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        af[mf] = *(const bf16x8*)&pimg[fa_qk_off(wmq + mf * 16 + lrow,
                                                 chunk)];
      int rbase = kk * 32 + (ll >> 4) * 8 + ((ll & 15) >> 2);
      int cb = ((ll & 15) & 3) * 8;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int cbyte = (wno + nf * 16) * 2 + cb;
        auto pv = (__attribute__((address_space(3))) bf16x4*)(
            (__attribute__((address_space(3))) char*)&vimg[buf][0] +
            fa_tr128_byte(rbase, cbyte));
        bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(pv);
        auto pv2 = (__attribute__((address_space(3))) bf16x4*)(
            (__attribute__((address_space(3))) char*)&vimg[buf][0] +
            fa_tr128_byte(rbase + 4, cbyte));
        bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(pv2);
        bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc_o[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc_o[mf][nf], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue ----
  bf16* obase = O + (long)b * Tq * hd + (long)h * D;
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = wmq + mf * 16 + (ll >> 4) * 4 + i;
      if (qb * 128 + row >= Tq) continue;
      float inv = l_i[mf][i] > 0.f ? 1.f / l_i[mf][i] : 0.f;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int col = wno + nf * 16 + (ll & 15);
        obase[(long)(qb * 128 + row) * hd + col] =
            f2bf(acc_o[mf][nf][i] * inv);
      }
    }
  }
  if ((ll & 15) == 0 && (lw >> 2) == 0) {
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wmq + mf * 16 + (ll >> 4) * 4 + i;
        if (qb * 128 + row >= Tq) continue;
        Ms[(long)z * Tq + qb * 128 + row] = m_i[mf][i];
        Ls[(long)z * Tq + qb * 128 + row] = l_i[mf][i];
      }
  }
}

__global__ __launch_bounds__(512, 1) void flash_fwd_core(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O,
    float* __restrict__ Ms, float* __restrict__ Ls, int B, int H, int Tq,
    int Tk, float alpha, int causal, const u64* __restrict__ seed_ptr,
    u64 salt, float pdrop, float dscale) {
  constexpr int D = 64;
  __shared__ __align__(16) bf16 qimg[128 * D];
  __shared__ __align__(16) bf16 kimg[2][128 * D];
  __shared__ __align__(16) bf16 vimg[2][128 * D];
  __shared__ __align__(16) bf16 pimg[128 * 128];
  __shared__ float sred[2][128][2];  // [stat m/l][row][tk-half wave]

  const int z = blockIdx.z;          // b*H + h
  const int b = z / H, h = z % H;
  const int qb = blockIdx.x;         // query block
  const long hd = (long)H * D;
  const int t = threadIdx.x;
  const int lw = t / WAVE, ll = t % WAVE;

  const u64 seed = seed_ptr ? seed_finalize(*seed_ptr + salt) : 0;
  const u32 thresh = (u32)(pdrop * 4294967296.0f);

  // ---- stage Q once: rows qb*128 + r, k = d (glds lane-linear, source
  // chunk XOR; geometry identical to the 512-thread TN kernel) ----
  const int srow = (((t >> 6) * 2) << 3) + ((t & 63) >> 3);  // pass-0 row
  const int scg = (ll & 7) ^ ((ll >> 3) & 7);
  {
    const bf16* qbase = Q + (long)b * Tq * hd + (long)h * D;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      int row = srow + p * 8;
      unsigned lbase = __builtin_amdgcn_readfirstlane(
          (unsigned)((lw * 2 + p) * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&qimg[0] + lbase);
      // tail blocks: clamp the source row (finite garbage; the scores,
      // stats and O writes of rows >= Tq are guarded/discarded)
      int qrow = min(qb * 128 + row, Tq - 1);
      const bf16* gp = qbase + (long)qrow * hd + scg * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds,
          16, 0, 0);
    }
  }

  // K / V chunk staging (kv chunk c -> rows c*128 + row)
  const bf16* kbase = K + (long)b * Tk * hd + (long)h * D;
  const bf16* vbase = V + (long)b * Tk * hd + (long)h * D;
  auto stage_kv = [&](int buf, int c) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      int row = srow + p * 8;
      unsigned lbase = __builtin_amdgcn_readfirstlane(
          (unsigned)((lw * 2 + p) * 1024));
      auto ldsK = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&kimg[buf][0] + lbase);
      auto ldsV = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)&vimg[buf][0] + lbase);
      int krow = min(c * 128 + row, Tk - 1);  // tail clamp (masked below)
      const bf16* gk = kbase + (long)krow * hd + scg * 8;
      // V tr image: source chunk key is (row&3)<<1 on 8-bf16 chunks —
      // row&3 here = (ll>>3)&3, chunk pos = ll&7
      int vcg = (ll & 7) ^ (((ll >> 3) & 3) << 1);
      const bf16* gv = vbase + (long)krow * hd + vcg * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gk, ldsK,
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gv, ldsV,
          16, 0, 0);
    }
  };

  const int wmq = (lw & 3) * 32;     // this wave's 32 query rows
  const int wnt = (lw >> 2) * 64;    // S phase: tk half
  const int wno = (lw >> 2) * 32;    // O phase: d half
  const int lrow = ll & 15, lk = ll >> 4;

  // persistent per-lane state: rows wmq + mf*16 + (ll>>4)*4 + i
  float m_i[2][4], l_i[2][4];
  f32x4 acc_o[2][2];
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      m_i[mf][i] = -3.0e38f;
      l_i[mf][i] = 0.f;
    }
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) acc_o[mf][nf] = f32x4{0, 0, 0, 0};

  const int nchunks_all = (Tk + 127) / 128;
  const int nchunks = causal ? min(qb + 1, nchunks_all) : nchunks_all;

  stage_kv(0, 0);
  __syncthreads();

  for (int c = 0; c < nchunks; ++c) {
    int buf = c & 1;
    if (c + 1 < nchunks) stage_kv(buf ^ 1, c + 1);

    // ---- S = alpha * Q K^T for this chunk (wave: 32 q x 64 tk) ----
    f32x4 acc_s[2][4];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) acc_s[mf][nf] = f32x4{0, 0, 0, 0};
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[2], bfr[4];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        af[mf] = *(const bf16x8*)&qimg[fa_qk_off(wmq + mf * 16 + lrow,
                                                 chunk)];
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        bfr[nf] = *(const bf16x8*)&kimg[buf][fa_qk_off(
            wnt + nf * 16 + lrow, chunk)];
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc_s[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc_s[mf][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }

    // ---- scale + causal mask + chunk row-max ----
    float cmax[2][4];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) cmax[mf][i] = -3.0e38f;
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int col = c * 128 + wnt + nf * 16 + (ll & 15);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int row = qb * 128 + wmq + mf * 16 + (ll >> 4) * 4 + i;
          float v = acc_s[mf][nf][i] * alpha;
          if ((causal && col > row) || col >= Tk) v = -3.0e38f;
          acc_s[mf][nf][i] = v;
          cmax[mf][i] = fmaxf(cmax[mf][i], v);
        }
      }
    // reduce over the 16 column lanes (rows replicated in lane groups)
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float v = cmax[mf][i];
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          v = fmaxf(v, __shfl_xor(v, off, 64));
        cmax[mf][i] = v;
      }
    // cross tk-half waves via LDS
    if ((ll & 15) == 0) {
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int i = 0; i < 4; ++i)
          sred[0][wmq + mf * 16 + (ll >> 4) * 4 + i][lw >> 2] =
              cmax[mf][i];
    }
    FA_LDS_BARRIER();
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int r = wmq + mf * 16 + (ll >> 4) * 4 + i;
        cmax[mf][i] = fmaxf(sred[0][r][0], sred[0][r][1]);
      }

    // ---- online rescale + P = exp(s - m_new) + chunk row-sum ----
    float csum[2][4];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float m_new = fmaxf(m_i[mf][i], cmax[mf][i]);
        float f = __expf(m_i[mf][i] - m_new);
        m_i[mf][i] = m_new;
        l_i[mf][i] *= f;
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) acc_o[mf][nf][i] *= f;
        csum[mf][i] = 0.f;
      }
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          float p = __expf(acc_s[mf][nf][i] - m_i[mf][i]);
          acc_s[mf][nf][i] = p;
          csum[mf][i] += p;
        }
      }
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float v = csum[mf][i];
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          v += __shfl_xor(v, off, 64);
        csum[mf][i] = v;
      }
    if ((ll & 15) == 0) {
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int i = 0; i < 4; ++i)
          sred[1][wmq + mf * 16 + (ll >> 4) * 4 + i][lw >> 2] =
              csum[mf][i];
    }
    FA_LDS_BARRIER();
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int r = wmq + mf * 16 + (ll >> 4) * 4 + i;
        l_i[mf][i] += sred[1][r][0] + sred[1][r][1];
      }

    // ---- dropout (unstored; bwd replays the hash) + write P image ----
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        int col = wnt + nf * 16 + (ll & 15);
        int gcol = c * 128 + col;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int row = wmq + mf * 16 + (ll >> 4) * 4 + i;
          float p = acc_s[mf][nf][i];
          if (thresh) {
            long grow = (long)qb * 128 + row;
            u32 r = hash_u32(seed, ((long)z * Tq + grow) * Tk + gcol);
            p = r >= thresh ? p * dscale : 0.f;
          }
          *(bf16*)((char*)pimg + fa_p_byte(row, col >> 3) +
                   (col & 7) * 2) = f2bf(p);
        }
      }
    FA_LDS_BARRIER();

    // ---- acc_o += P~ V (wave: 32 q x 32 d), K dim = 128 tk ----
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      bf16x8 af[2], bfr[2];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        af[mf] = *(const bf16x8*)((const char*)pimg +
                                  fa_p_byte(wmq + mf * 16 + lrow, chunk));
      int rbase = kk * 32 + (ll >> 4) * 8 + ((ll & 15) >> 2);
      int cb = ((ll & 15) & 3) * 8;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int cbyte = (wno + nf * 16) * 2 + cb;
        auto pv = (__attribute__((address_space(3))) bf16x4*)(
            (__attribute__((address_space(3))) char*)&vimg[buf][0] +
            fa_v_byte(rbase, cbyte));
        bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(pv);
        auto pv2 = (__attribute__((address_space(3))) bf16x4*)(
            (__attribute__((address_space(3))) char*)&vimg[buf][0] +
            fa_v_byte(rbase + 4, cbyte));
        bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(pv2);
        bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc_o[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc_o[mf][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();  // pimg reused; kv buffers swap — and the chunk's
                      // prefetch (issued a full iteration ago) drains
  }

  // ---- epilogue: O = acc_o / l, write strided [B, Tq, H*D]; stats ----
  bf16* obase = O + (long)b * Tq * hd + (long)h * D;
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = wmq + mf * 16 + (ll >> 4) * 4 + i;
      if (qb * 128 + row >= Tq) continue;  // tail guard
      float inv = l_i[mf][i] > 0.f ? 1.f / l_i[mf][i] : 0.f;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int col = wno + nf * 16 + (ll & 15);
        obase[(long)(qb * 128 + row) * hd + col] =
            f2bf(acc_o[mf][nf][i] * inv);
      }
    }
  }
  if ((ll & 15) == 0 && (lw >> 2) == 0) {
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wmq + mf * 16 + (ll >> 4) * 4 + i;
        if (qb * 128 + row >= Tq) continue;  // tail guard
        Ms[(long)z * Tq + qb * 128 + row] = m_i[mf][i];
        Ls[(long)z * Tq + qb * 128 + row] = l_i[mf][i];
      }
  }
}

void flash_fwd_bf16(const bf16* Q, const bf16* K, const bf16* V, bf16* O,
                    float* Ms, float* Ls, int B, int H, int Tq, int Tk,
                    int D, float alpha, bool causal, const u64* seed_ptr,
                    u64 salt, float pdrop, hipStream_t stream) {
  dim3 grid((Tq + 127) / 128, 1, B * H);
  float dscale = pdrop > 0.f ? 1.f / (1.f - pdrop) : 1.f;
  if (D == 128) {
    flash_fwd_core_d128<<<grid, 512, 0, stream>>>(
        Q, K, V, O, Ms, Ls, B, H, Tq, Tk, alpha, causal ? 1 : 0,
        pdrop > 0.f ? seed_ptr : nullptr, salt, pdrop, dscale);
    return;
  }
  flash_fwd_core<<<grid, 512, 0, stream>>>(
      Q, K, V, O, Ms, Ls, B, H, Tq, Tk, alpha, causal ? 1 : 0,
      pdrop > 0.f ? seed_ptr : nullptr, salt, pdrop, dscale);
}

// ---------------------------------------------------------------------------
// Backward.  Grid (Tk/128 kv-blocks, 1, B*H): each block owns dK/dV for
// its 128 kv rows (register accumulators across the q-chunk loop); dQ is
// accumulated into a global f32 buffer with atomics.  Probabilities and
// the dropout mask are RECOMPUTED from the saved (m, l) stats and the
// RNG seed — S/P/mask never hit HBM in either direction.
//
// LDS is a managed arena (146 KiB peak): the Q-TN/dO-TN images are dead
// after the two TN GEMMs and their 32 KiB is reused for the dS tr-image.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(512, 1) void flash_bwd_core(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ Ms, const float* __restrict__ Ls,
    const float* __restrict__ Drow, float* __restrict__ dQacc,
    bf16* __restrict__ dKout, bf16* __restrict__ dVout, int B, int H,
    int Tq, int Tk, float alpha, int causal,
    const u64* __restrict__ seed_ptr, u64 salt, float pdrop,
    float dscale) {
  constexpr int D = 64;
  // q processed in 64-row chunks so no LDS region needs to alias:
  // K/V resident (48 KiB) + per-chunk q/dO images (32 KiB) + pd/ds
  // images (32 KiB) = 112 KiB.
  __shared__ __align__(16) bf16 ktn[128 * D];
  __shared__ __align__(16) bf16 ktr[128 * D];
  __shared__ __align__(16) bf16 vtn[128 * D];
  __shared__ __align__(16) bf16 qtn[64 * D];
  __shared__ __align__(16) bf16 qtr[64 * D];
  __shared__ __align__(16) bf16 dotn[64 * D];
  __shared__ __align__(16) bf16 dotr[64 * D];
  __shared__ __align__(16) bf16 ptn[128 * 64];   // pd, [tk][64 q]
  __shared__ __align__(16) bf16 dstr[128 * 64];  // ds, tr image [tk][64 q]
  __shared__ float stats[3][64];                 // m, l, Drow per q-chunk

  const int z = blockIdx.z;
  const int b = z / H, h = z % H;
  const int kvb = blockIdx.x;
  const long hd = (long)H * D;
  const int t = threadIdx.x;
  const int lw = t / WAVE, ll = t % WAVE;

  const u64 seed = seed_ptr ? seed_finalize(*seed_ptr + salt) : 0;
  const u32 thresh = (u32)(pdrop * 4294967296.0f);

  const int srow = (((t >> 6) * 2) << 3) + ((t & 63) >> 3);  // 128-row
  const int srow64 = ((t >> 6) << 3) + ((t & 63) >> 3);      // 64-row
  const int scg = (ll & 7) ^ ((ll >> 3) & 7);
  const int tcg = (ll & 7) ^ (((ll >> 3) & 3) << 1);

  auto stage128 = [&](bf16* img, const bf16* base, int row0, int cg,
                      int rmax) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      int row = srow + p * 8;
      unsigned lbase = __builtin_amdgcn_readfirstlane(
          (unsigned)((lw * 2 + p) * 1024));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)img + lbase);
      const bf16* gp = base + (long)min(row0 + row, rmax - 1) * hd + cg * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds,
          16, 0, 0);
    }
  };
  auto stage64r = [&](bf16* img, const bf16* base, int row0, int cg,
                      int rmax) {
    unsigned lbase =
        __builtin_amdgcn_readfirstlane((unsigned)(lw * 1024));
    auto lds = (__attribute__((address_space(3))) unsigned int*)(
        (__attribute__((address_space(3))) char*)img + lbase);
    const bf16* gp = base + (long)min(row0 + srow64, rmax - 1) * hd +
                     cg * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
        0, 0);
  };

  const bf16* qbase = Q + (long)b * Tq * hd + (long)h * D;
  const bf16* kbase = K + (long)b * Tk * hd + (long)h * D;
  const bf16* vbase = V + (long)b * Tk * hd + (long)h * D;
  const bf16* dobase = dO + (long)b * Tq * hd + (long)h * D;

  stage128(ktn, kbase, kvb * 128, scg, Tk);
  stage128(ktr, kbase, kvb * 128, tcg, Tk);
  stage128(vtn, vbase, kvb * 128, scg, Tk);

  const int wmq = (lw & 3) * 32;   // 32 tk rows of this wave
  const int wnt = (lw >> 2) * 32;  // 32 q cols (of the 64-chunk)
  const int wno = (lw >> 2) * 32;  // 32 d cols
  const int lrow = ll & 15, lk = ll >> 4;

  f32x4 acc_dv[2][2], acc_dk[2][2];
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      acc_dv[mf][nf] = f32x4{0, 0, 0, 0};
      acc_dk[mf][nf] = f32x4{0, 0, 0, 0};
    }

  const int qc0 = causal ? kvb * 2 : 0;
  const int nq = (Tq + 63) / 64;

  for (int qc = qc0; qc < nq; ++qc) {
    stage64r(qtn, qbase, qc * 64, scg, Tq);
    stage64r(qtr, qbase, qc * 64, tcg, Tq);
    stage64r(dotn, dobase, qc * 64, scg, Tq);
    stage64r(dotr, dobase, qc * 64, tcg, Tq);
    for (int i = t; i < 64; i += 512) {
      long qg = (long)z * Tq + min(qc * 64 + i, Tq - 1);
      stats[0][i] = Ms[qg];
      stats[1][i] = Ls[qg];
      stats[2][i] = Drow[qg];
    }
    __syncthreads();

    // ---- S_T = K Q^T and dP_T = V dO^T ([128 tk][64 q]) ----
    f32x4 acc_st[2][2], acc_dp[2][2];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        acc_st[mf][nf] = f32x4{0, 0, 0, 0};
        acc_dp[mf][nf] = f32x4{0, 0, 0, 0};
      }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 ak[2], bq[2];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        ak[mf] = *(const bf16x8*)&ktn[fa_qk_off(wmq + mf * 16 + lrow,
                                                chunk)];
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        bq[nf] = *(const bf16x8*)&qtn[fa_qk_off(wnt + nf * 16 + lrow,
                                                chunk)];
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc_st[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              ak[mf], bq[nf], acc_st[mf][nf], 0, 0, 0);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 av[2], bdo[2];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        av[mf] = *(const bf16x8*)&vtn[fa_qk_off(wmq + mf * 16 + lrow,
                                                chunk)];
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        bdo[nf] = *(const bf16x8*)&dotn[fa_qk_off(wnt + nf * 16 + lrow,
                                                  chunk)];
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc_dp[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              av[mf], bdo[nf], acc_dp[mf][nf], 0, 0, 0);
    }

    // ---- p, pd, ds; pd -> ptn (TN image), ds -> dstr (tr image) ----
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int qcol = wnt + nf * 16 + (ll & 15);
        long qg = (long)qc * 64 + qcol;
        float m = stats[0][qcol], linv = stats[1][qcol];
        linv = linv > 0.f ? 1.f / linv : 0.f;
        float dr = stats[2][qcol];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int tkrow = wmq + mf * 16 + (ll >> 4) * 4 + i;
          long kg = (long)kvb * 128 + tkrow;
          float sv = acc_st[mf][nf][i] * alpha;
          float pv = ((causal && kg > qg) || qg >= Tq || kg >= Tk)
                         ? 0.f
                         : __expf(sv - m) * linv;
          float keep = 1.f;
          if (thresh) {
            u32 r = hash_u32(seed, ((long)z * Tq + qg) * Tk + kg);
            keep = r >= thresh ? dscale : 0.f;
          }
          float pd = pv * keep;
          float dp = acc_dp[mf][nf][i] * keep;
          float ds = pv * (dp - dr) * alpha;
          *(bf16*)((char*)ptn + fa_qk_off(tkrow, qcol >> 3) * 2 +
                   (qcol & 7) * 2) = f2bf(pd);
          *(bf16*)((char*)dstr + fa_v_byte(tkrow, qcol * 2)) = f2bf(ds);
        }
      }
    __syncthreads();

    // ---- dV += pd · dO (K dim = 64 q) ----
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[2], bfr[2];
      int chunk = kk * 4 + lk;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        af[mf] = *(const bf16x8*)&ptn[fa_qk_off(wmq + mf * 16 + lrow,
                                                chunk)];
      int rbase = kk * 32 + (ll >> 4) * 8 + ((ll & 15) >> 2);
      int cb = ((ll & 15) & 3) * 8;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int cbyte = (wno + nf * 16) * 2 + cb;
        auto p1 = (__attribute__((address_space(3))) bf16x4*)(
            (__attribute__((address_space(3))) char*)dotr +
            fa_v_byte(rbase, cbyte));
        auto p2 = (__attribute__((address_space(3))) bf16x4*)(
            (__attribute__((address_space(3))) char*)dotr +
            fa_v_byte(rbase + 4, cbyte));
        bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
        bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
        bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc_dv[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc_dv[mf][nf], 0, 0, 0);
    }

    // ---- dK += ds · Q (ds via b128 on the tr-keyed image) ----
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[2], bfr[2];
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
        int row = wmq + mf * 16 + lrow;
        int chunk = kk * 4 + lk;
        af[mf] = *(const bf16x8*)((const char*)dstr +
                                  fa_v_byte(row, chunk * 16));
      }
      int rbase = kk * 32 + (ll >> 4) * 8 + ((ll & 15) >> 2);
      int cb = ((ll & 15) & 3) * 8;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int cbyte = (wno + nf * 16) * 2 + cb;
        auto p1 = (__attribute__((address_space(3))) bf16x4*)(
            (__attribute__((address_space(3))) char*)qtr +
            fa_v_byte(rbase, cbyte));
        auto p2 = (__attribute__((address_space(3))) bf16x4*)(
            (__attribute__((address_space(3))) char*)qtr +
            fa_v_byte(rbase + 4, cbyte));
        bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
        bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
        bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc_dk[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc_dk[mf][nf], 0, 0, 0);
    }

    // ---- dQ partial: WG over tk (A = dstr tr image [tk][64 q],
    //      B = ktr [tk][64 d]); wave = 16 q x 32 d ----
    {
      f32x4 acc_dq[2];
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) acc_dq[nf] = f32x4{0, 0, 0, 0};
      const int wq16 = (lw & 3) * 16;  // Mo offset (64 q total)
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        bf16x8 af, bfr[2];
        int rbase = kk * 32 + (ll >> 4) * 8 + ((ll & 15) >> 2);
        int cb = ((ll & 15) & 3) * 8;
        {
          int cbyte = wq16 * 2 + cb;
          auto p1 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)dstr +
              fa_v_byte(rbase, cbyte));
          auto p2 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)dstr +
              fa_v_byte(rbase + 4, cbyte));
          bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
          bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
          af = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
        }
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
          int cbyte = (wno + nf * 16) * 2 + cb;
          auto p1 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)ktr +
              fa_v_byte(rbase, cbyte));
          auto p2 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)ktr +
              fa_v_byte(rbase + 4, cbyte));
          bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
          bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
          bfr[nf] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6,
                                            7);
        }
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc_dq[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bfr[nf], acc_dq[nf], 0, 0, 0);
      }
      float* dqb = dQacc + ((long)b * Tq + qc * 64) * hd + (long)h * D;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int col = wno + nf * 16 + (ll & 15);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int row = wq16 + (ll >> 4) * 4 + i;
          if (qc * 64 + row >= Tq) continue;  // tail guard
          atomicAdd(&dqb[(long)row * hd + col], acc_dq[nf][i]);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: dK, dV (strided bf16) ----
  bf16* dkb = dKout + ((long)b * Tk + kvb * 128) * hd + (long)h * D;
  bf16* dvb = dVout + ((long)b * Tk + kvb * 128) * hd + (long)h * D;
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      int col = wno + nf * 16 + (ll & 15);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wmq + mf * 16 + (ll >> 4) * 4 + i;
        if (kvb * 128 + row >= Tk) continue;  // tail guard
        dkb[(long)row * hd + col] = f2bf(acc_dk[mf][nf][i]);
        dvb[(long)row * hd + col] = f2bf(acc_dv[mf][nf][i]);
      }
    }
}

// rowwise dot for the backward: Drow[z][q] = sum_d dO * O  (wave-per-row)
__global__ __launch_bounds__(256) void flash_drow(
    const bf16* __restrict__ dO, const bf16* __restrict__ O,
    float* __restrict__ Drow, int B, int H, int Tq, int D,
    float* __restrict__ dqzero, long nzero4) {
  // fused dQacc zeroing (float4 grid-stride) — replaces the separate
  // at::zeros fill the backward otherwise pays before its atomics
  if (dqzero) {
    auto z4 = (f32x4*)dqzero;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nzero4;
         i += (long)gridDim.x * blockDim.x)
      z4[i] = f32x4{0.f, 0.f, 0.f, 0.f};
  }
  const long hd = (long)H * D;
  const int w = threadIdx.x >> 6, l = threadIdx.x & 63;
  long rows = (long)B * H * Tq;
  const int nvec = D / 8;  // bf16x8 per row (8 for D=64, 16 for D=128)
  for (long r = blockIdx.x * 4 + w; r < rows; r += 4L * gridDim.x) {
    int z = (int)(r / Tq);
    int q = (int)(r % Tq);
    int b = z / H, h = z % H;
    const bf16* dop = dO + ((long)b * Tq + q) * hd + (long)h * D;
    const bf16* op = O + ((long)b * Tq + q) * hd + (long)h * D;
    float acc = 0.f;
    for (int v = l; v < nvec; v += 64) {
      bf16x8 a = *(const bf16x8*)&dop[v * 8];
      bf16x8 c = *(const bf16x8*)&op[v * 8];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += bf2f(a[j]) * bf2f(c[j]);
    }
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) acc += __shfl_xor(acc, off, 64);
    if (l == 0) Drow[r] = acc;
  }
}

// ---------------------------------------------------------------------------
// D = 128 backward: kv-block 128, q processed in 32-row chunks (LDS:
// K/Ktr/V 3x32K + q/dO images 4x8K + p/ds 2x8K = 144 KiB).  The narrow
// (32-wide) p/ds images are PLAIN row-major: their 64-B rows put the 4
// rows of a tr read 16 banks apart — conflict-free without a swizzle.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(512, 1) void flash_bwd_core_d128(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ Ms, const float* __restrict__ Ls,
    const float* __restrict__ Drow, float* __restrict__ dQacc,
    bf16* __restrict__ dKout, bf16* __restrict__ dVout, int B, int H,
    int Tq, int Tk, float alpha, int causal,
    const u64* __restrict__ seed_ptr, u64 salt, float pdrop,
    float dscale) {
  constexpr int D = 128;
  __shared__ __align__(16) bf16 ktn[128 * D];
  __shared__ __align__(16) bf16 ktr[128 * D];
  __shared__ __align__(16) bf16 vtn[128 * D];
  __shared__ __align__(16) bf16 qtn[32 * D];
  __shared__ __align__(16) bf16 qtr[32 * D];
  __shared__ __align__(16) bf16 dotn[32 * D];
  __shared__ __align__(16) bf16 dotr[32 * D];
  __shared__ __align__(16) bf16 ptn[128 * 32];   // pd, [tk][32 q], plain
  __shared__ __align__(16) bf16 dstr[128 * 32];  // ds, [tk][32 q], plain
  __shared__ float stats[3][32];

  const int z = blockIdx.z;
  const int b = z / H, h = z % H;
  const int kvb = blockIdx.x;
  const long hd = (long)H * D;
  const int t = threadIdx.x;
  const int lw = t / WAVE, ll = t % WAVE;

  const u64 seed = seed_ptr ? seed_finalize(*seed_ptr + salt) : 0;
  const u32 thresh = (u32)(pdrop * 4294967296.0f);

  // 256-B-row staging (32 rows per glds pass; key row&15 TN, (row&3)<<1 tr)
  const int srow4 = lw * 4 + (ll >> 4);
  const int cg_tn = (ll & 15) ^ (srow4 & 15);
  const int cg_tr = (ll & 15) ^ ((srow4 & 3) << 1);

  auto stage128r = [&](bf16* img, const bf16* base, int row0, int cg,
                       int rmax) {
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      int row = p * 32 + srow4;
      unsigned lbase = __builtin_amdgcn_readfirstlane(
          (unsigned)((p * 32 + lw * 4) * 256));
      auto lds = (__attribute__((address_space(3))) unsigned int*)(
          (__attribute__((address_space(3))) char*)img + lbase);
      const bf16* gp = base + (long)min(row0 + row, rmax - 1) * hd + cg * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp, lds,
          16, 0, 0);
    }
  };
  auto stage32r = [&](bf16* img, const bf16* base, int row0, int cg,
                      int rmax) {
    unsigned lbase =
        __builtin_amdgcn_readfirstlane((unsigned)(lw * 4 * 256));
    auto lds = (__attribute__((address_space(3))) unsigned int*)(
        (__attribute__((address_space(3))) char*)img + lbase);
    const bf16* gp =
        base + (long)min(row0 + srow4, rmax - 1) * hd + cg * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gp, lds, 16,
        0, 0);
  };

  const bf16* qbase = Q + (long)b * Tq * hd + (long)h * D;
  const bf16* kbase = K + (long)b * Tk * hd + (long)h * D;
  const bf16* vbase = V + (long)b * Tk * hd + (long)h * D;
  const bf16* dobase = dO + (long)b * Tq * hd + (long)h * D;

  stage128r(ktn, kbase, kvb * 128, cg_tn, Tk);
  stage128r(ktr, kbase, kvb * 128, cg_tr, Tk);
  stage128r(vtn, vbase, kvb * 128, cg_tn, Tk);

  const int wmq = (lw & 3) * 32;   // 32 tk rows of this wave
  const int wnt = (lw >> 2) * 16;  // 16 q cols (of the 32-chunk)
  const int wno = (lw >> 2) * 64;  // 64 d cols (of 128)
  const int lrow = ll & 15, lk = ll >> 4;

  f32x4 acc_dv[2][4], acc_dk[2][4];
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      acc_dv[mf][nf] = f32x4{0, 0, 0, 0};
      acc_dk[mf][nf] = f32x4{0, 0, 0, 0};
    }

  const int qc0 = causal ? kvb * 4 : 0;
  const int nq = (Tq + 31) / 32;

  for (int qc = qc0; qc < nq; ++qc) {
    stage32r(qtn, qbase, qc * 32, cg_tn, Tq);
    stage32r(qtr, qbase, qc * 32, cg_tr, Tq);
    stage32r(dotn, dobase, qc * 32, cg_tn, Tq);
    stage32r(dotr, dobase, qc * 32, cg_tr, Tq);
    for (int i = t; i < 32; i += 512) {
      long qg = (long)z * Tq + min(qc * 32 + i, Tq - 1);
      stats[0][i] = Ms[qg];
      stats[1][i] = Ls[qg];
      stats[2][i] = Drow[qg];
    }
    __syncthreads();

    // ---- S_T = K Q^T and dP_T = V dO^T ([128 tk][32 q]) ----
    f32x4 acc_st[2], acc_dp[2];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
      acc_st[mf] = f32x4{0, 0, 0, 0};
      acc_dp[mf] = f32x4{0, 0, 0, 0};
    }
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      int chunk = kk * 4 + lk;
      bf16x8 bq = *(const bf16x8*)&qtn[fa_qk128_off(wnt + lrow, chunk)];
      bf16x8 bdo =
          *(const bf16x8*)&dotn[fa_qk128_off(wnt + lrow, chunk)];
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
        bf16x8 ak = *(const bf16x8*)&ktn[fa_qk128_off(
            wmq + mf * 16 + lrow, chunk)];
        bf16x8 av = *(const bf16x8*)&vtn[fa_qk128_off(
            wmq + mf * 16 + lrow, chunk)];
        acc_st[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            ak, bq, acc_st[mf], 0, 0, 0);
        acc_dp[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            av, bdo, acc_dp[mf], 0, 0, 0);
      }
    }

    // ---- p, pd, ds -> plain [128 tk][32 q] images ----
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
      int qcol = wnt + (ll & 15);
      long qg = (long)qc * 32 + qcol;
      float m = stats[0][qcol], linv = stats[1][qcol];
      linv = linv > 0.f ? 1.f / linv : 0.f;
      float dr = stats[2][qcol];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int tkrow = wmq + mf * 16 + (ll >> 4) * 4 + i;
        long kg = (long)kvb * 128 + tkrow;
        float sv = acc_st[mf][i] * alpha;
        float pv = ((causal && kg > qg) || qg >= Tq || kg >= Tk)
                       ? 0.f
                       : __expf(sv - m) * linv;
        float keep = 1.f;
        if (thresh) {
          u32 r = hash_u32(seed, ((long)z * Tq + qg) * Tk + kg);
          keep = r >= thresh ? dscale : 0.f;
        }
        float pd = pv * keep;
        float dp = acc_dp[mf][i] * keep;
        float ds = pv * (dp - dr) * alpha;
        ptn[tkrow * 32 + qcol] = f2bf(pd);
        dstr[tkrow * 32 + qcol] = f2bf(ds);
      }
    }
    __syncthreads();

    // ---- dV += pd . dO  and  dK += ds . Q (K dim = 32 q) ----
    {
      int rbase = (ll >> 4) * 8 + ((ll & 15) >> 2);
      int cb = ((ll & 15) & 3) * 8;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
        int row = wmq + mf * 16 + lrow;
        bf16x8 apd = *(const bf16x8*)&ptn[row * 32 + lk * 8];
        bf16x8 ads = *(const bf16x8*)&dstr[row * 32 + lk * 8];
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) {
          int cbyte = (wno + nf * 16) * 2 + cb;
          auto p1 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)dotr +
              fa_tr128_byte(rbase, cbyte));
          auto p2 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)dotr +
              fa_tr128_byte(rbase + 4, cbyte));
          bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
          bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
          bf16x8 bdo =
              __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
          acc_dv[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              apd, bdo, acc_dv[mf][nf], 0, 0, 0);
          auto p3 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)qtr +
              fa_tr128_byte(rbase, cbyte));
          auto p4 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)qtr +
              fa_tr128_byte(rbase + 4, cbyte));
          bf16x4 lo2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p3);
          bf16x4 hi2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p4);
          bf16x8 bq =
              __builtin_shufflevector(lo2, hi2, 0, 1, 2, 3, 4, 5, 6, 7);
          acc_dk[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              ads, bq, acc_dk[mf][nf], 0, 0, 0);
        }
      }
    }

    // ---- dQ partial: [32 q][128 d] over tk = 128 (wave: 16 q x 32 d) --
    {
      f32x4 acc_dq[2];
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) acc_dq[nf] = f32x4{0, 0, 0, 0};
      const int wq16 = (lw & 1) * 16;   // q offset (2 wave groups x 16)
      const int wd32 = (lw >> 1) * 32;  // d offset (4 wave groups x 32)
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        int rbase = kk * 32 + (ll >> 4) * 8 + ((ll & 15) >> 2);
        int cb = ((ll & 15) & 3) * 8;
        bf16x8 af;
        {
          // ds tr read from the plain [128 tk][32 q] image: 64-B rows
          int cbyte = wq16 * 2 + cb;
          auto p1 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)dstr +
              rbase * 64 + cbyte);
          auto p2 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)dstr +
              (rbase + 4) * 64 + cbyte);
          bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
          bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
          af = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
        }
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
          int cbyte = (wd32 + nf * 16) * 2 + cb;
          auto p1 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)ktr +
              fa_tr128_byte(rbase, cbyte));
          auto p2 = (__attribute__((address_space(3))) bf16x4*)(
              (__attribute__((address_space(3))) char*)ktr +
              fa_tr128_byte(rbase + 4, cbyte));
          bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
          bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
          bf16x8 bk =
              __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
          acc_dq[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bk, acc_dq[nf], 0, 0, 0);
        }
      }
      float* dqb = dQacc + ((long)b * Tq + qc * 32) * hd + (long)h * D;
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int col = wd32 + nf * 16 + (ll & 15);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int row = wq16 + (ll >> 4) * 4 + i;
          if (qc * 32 + row >= Tq) continue;
          atomicAdd(&dqb[(long)row * hd + col], acc_dq[nf][i]);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: dK, dV ([128 tk][128 d] strided bf16) ----
  bf16* dkb = dKout + ((long)b * Tk + kvb * 128) * hd + (long)h * D;
  bf16* dvb = dVout + ((long)b * Tk + kvb * 128) * hd + (long)h * D;
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      int col = wno + nf * 16 + (ll & 15);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wmq + mf * 16 + (ll >> 4) * 4 + i;
        if (kvb * 128 + row >= Tk) continue;
        dkb[(long)row * hd + col] = f2bf(acc_dk[mf][nf][i]);
        dvb[(long)row * hd + col] = f2bf(acc_dv[mf][nf][i]);
      }
    }
}

void flash_bwd_bf16(const bf16* Q, const bf16* K, const bf16* V,
                    const bf16* O, const bf16* dOp, const float* Ms,
                    const float* Ls, float* Drow, float* dQacc, bf16* dK,
                    bf16* dV, int B, int H, int Tq, int Tk, int D,
                    float alpha, bool causal, const u64* seed_ptr,
                    u64 salt, float pdrop, hipStream_t stream) {
  long rows = (long)B * H * Tq;
  int blocks = (int)min((rows + 3) / 4, 2048L);
  flash_drow<<<blocks, 256, 0, stream>>>(dOp, O, Drow, B, H, Tq, D, dQacc,
                                         (long)B * Tq * H * D / 4);
  float dscale = pdrop > 0.f ? 1.f / (1.f - pdrop) : 1.f;
  dim3 grid((Tk + 127) / 128, 1, B * H);
  if (D == 128) {
    flash_bwd_core_d128<<<grid, 512, 0, stream>>>(
        Q, K, V, dOp, Ms, Ls, Drow, dQacc, dK, dV, B, H, Tq, Tk, alpha,
        causal ? 1 : 0, pdrop > 0.f ? seed_ptr : nullptr, salt, pdrop,
        dscale);
    return;
  }
  flash_bwd_core<<<grid, 512, 0, stream>>>(
      Q, K, V, dOp, Ms, Ls, Drow, dQacc, dK, dV, B, H, Tq, Tk, alpha,
      causal ? 1 : 0, pdrop > 0.f ? seed_ptr : nullptr, salt, pdrop,
      dscale);
}

}  // namespace mxa
