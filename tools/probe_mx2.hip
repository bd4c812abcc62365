// MX scale lane-mapping probe: which lane-group's scale byte applies to
// which k-block of a row?  Data = 1.0 only inside ONE k-block (bi);
// A-scale byte0 of lane l = 127 + (l>>4).  D[m][n]/32 = 2^{g} reveals
// the lane-group g whose scale the HW applied to block bi.
// Also E0: all-ones data, per-lane i32 = 4 distinct bytes -> how many
// bytes are consumed.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void k(const unsigned char* A, const int* Asc,
                  const unsigned char* B, float* D) {
  int l = threadIdx.x;
  i32x8 a, b;
  const unsigned char* ap = A + (l & 15) * 128 + (l >> 4) * 32;
  const unsigned char* bp = B + (l & 15) * 128 + (l >> 4) * 32;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = *(const int*)(ap + i * 4);
    b[i] = *(const int*)(bp + i * 4);
  }
  int sa = Asc[l];
  int sb = 127 | (127 << 8) | (127 << 16) | (127 << 24);
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(a, b, c, 0, 0, 0,
                                                       sa, 0, sb);
#pragma unroll
  for (int i = 0; i < 4; ++i)
    D[((l >> 4) * 4 + i) * 16 + (l & 15)] = c[i];
}

int main() {
  unsigned char hA[16 * 128], hB[16 * 128];
  int hAs[64];
  unsigned char *dA, *dB;
  int* dAs;
  float* dD;
  hipMalloc(&dA, sizeof(hA));
  hipMalloc(&dB, sizeof(hB));
  hipMalloc(&dAs, sizeof(hAs));
  hipMalloc(&dD, 256 * 4);
  float hD[256];
  auto run = [&]() {
    hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
    hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
    hipMemcpy(dAs, hAs, sizeof(hAs), hipMemcpyHostToDevice);
    hipLaunchKernelGGL(k, dim3(1), dim3(64), 0, 0, dA, dAs, dB, dD);
    hipDeviceSynchronize();
    hipMemcpy(hD, dD, sizeof(hD), hipMemcpyDeviceToHost);
  };
  for (int i = 0; i < 16 * 128; ++i) hB[i] = 0x38;  // 1.0

  // E0: all-ones data; every lane i32 = bytes (128,129,130,131)
  for (int i = 0; i < 16 * 128; ++i) hA[i] = 0x38;
  for (int l = 0; l < 64; ++l)
    hAs[l] = 128 | (129 << 8) | (130 << 16) | (131 << 24);
  run();
  printf("E0 D[0][0]/32 = %.2f (byte0-only->2*4=8; bytes0-3->2+4+8+16=30)\n",
         hD[0] / 32.0);

  // E1..E4: ones only in block bi; scale byte0 = 127 + (l>>4)
  for (int bi = 0; bi < 4; ++bi) {
    for (int m = 0; m < 16; ++m)
      for (int kk = 0; kk < 128; ++kk)
        hA[m * 128 + kk] = (kk / 32 == bi) ? 0x38 : 0;
    for (int l = 0; l < 64; ++l) {
      int s0 = 127 + (l >> 4);
      hAs[l] = s0 | (127 << 8) | (127 << 16) | (127 << 24);
    }
    run();
    printf("block %d: D[0][0]/32 = %.2f -> scale from lane-group %d\n",
           bi, hD[0] / 32.0, (int)round(log2(hD[0] / 32.0)));
  }
  return 0;
}
