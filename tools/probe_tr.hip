// Empirical semantics probe for ds_read_b64_tr_b16 on gfx950.
// Fills LDS with bf16 values encoding their element index, issues the
// transpose-read with per-lane addresses lane*8 (linear) and a second
// pattern, and dumps what each lane receives.  Build + run standalone:
//   hipcc --offload-arch=gfx950 -O3 probe_tr.hip -o probe_tr && ./probe_tr
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __bf16 bf16;

__global__ void probe(float* out, int mode) {
  __shared__ __align__(16) bf16 lds[1024];
  int t = threadIdx.x;
  for (int i = t; i < 1024; i += 64) lds[i] = (bf16)(float)i;
  __syncthreads();
  // per-lane byte address
  unsigned addr;
  if (mode == 0) addr = t * 8;                       // linear 8B segments
  else if (mode == 1) addr = (t & 15) * 8 + (t >> 4) * 128;
  else addr = ((t & 3) * 16 + ((t >> 2) & 3) * 4) * 2 + (t >> 4) * 128;
  unsigned lds_base = (unsigned)(unsigned long long)(
      (__attribute__((address_space(3))) char*)&lds[0]);
  unsigned vaddr = lds_base + addr;
  unsigned long long r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r) : "v"(vaddr));
  __builtin_amdgcn_sched_barrier(0);
  unsigned short* parts = (unsigned short*)&r;
  for (int j = 0; j < 4; ++j) {
    bf16 v = *(bf16*)&parts[j];
    out[(mode * 64 + t) * 4 + j] = (float)v;
  }
}

int main() {
  float* d;
  hipMalloc(&d, 3 * 64 * 4 * sizeof(float));
  hipMemset(d, 0, 3 * 64 * 4 * sizeof(float));
  for (int mode = 0; mode < 3; ++mode)
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
  hipDeviceSynchronize();
  float h[3 * 64 * 4];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  for (int mode = 0; mode < 3; ++mode) {
    printf("== mode %d: lane -> [elem indices of 4 received bf16]\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("l%02d: %5.0f %5.0f %5.0f %5.0f   ", l,
             h[(mode * 64 + l) * 4 + 0], h[(mode * 64 + l) * 4 + 1],
             h[(mode * 64 + l) * 4 + 2], h[(mode * 64 + l) * 4 + 3]);
      if (l % 2 == 1) printf("\n");
    }
  }
  return 0;
}
