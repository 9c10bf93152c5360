// Empirical layout probe for gfx950 ds_read_b64_tr_b16: fill LDS with the
// linear element index, issue the transpose-read at several address modes,
// and dump what each lane received. Standalone (no torch): hipcc
// --offload-arch=gfx950 tools/tr_probe.hip -o tools/tr_probe
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(unsigned short* out, int mode) {
  __shared__ __align__(16) unsigned short lds[2048];  // 4 KiB
  const int tid = threadIdx.x;
  for (int i = tid; i < 2048; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  unsigned int addr;
  switch (mode) {
    case 0: addr = (unsigned int)(unsigned long long)&lds[tid * 4]; break;     // lane*8 B
    case 1: addr = (unsigned int)(unsigned long long)&lds[(tid & 15) * 4 + (tid >> 4) * 64]; break;
    case 2: addr = (unsigned int)(unsigned long long)&lds[(tid & 15) + (tid >> 4) * 64]; break;
    default: addr = (unsigned int)(unsigned long long)&lds[(tid & 3) * 16 + (tid >> 2) * 64]; break;
  }
  unsigned long long v = 0;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr));
#pragma unroll
  for (int j = 0; j < 4; ++j) out[tid * 4 + j] = (unsigned short)(v >> (16 * j));
}

int main() {
  unsigned short* d;
  (void)hipMalloc(&d, 64 * 4 * sizeof(unsigned short));
  unsigned short h[256];
  for (int mode = 0; mode < 4; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("== mode %d (addr of lane l -> elements received) ==\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("l%02d: %4d %4d %4d %4d%s", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3],
             (l % 4 == 3) ? "\n" : "   ");
    }
  }
  (void)hipFree(d);
  return 0;
}
