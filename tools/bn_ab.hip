// A/B probe for the BN element-pass bandwidth (bn_bwd2-shaped triad:
// read dy, read x, read relu_mask, write dx). Production kernel lives in
// active_learning_amd/ops/hip/bn.hip; this file only exists to measure
// structural variants against a pure-streaming roofline on gfx950.
// Build: hipcc --offload-arch=gfx950 -O3 -shared -fPIC -o bn_ab.so bn_ab.hip
#include "../active_learning_amd/ops/hip/al_common.h"

// ---- v0: production structure (fixed channel strip, #pragma unroll 2) ----
__global__ void bwd_v0(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                       const unsigned char* __restrict__ mask,
                       const float* __restrict__ par, float inv_n,
                       bf16* __restrict__ dx, long rows, int C, int cg_per_block) {
  const int cg_local = threadIdx.x % cg_per_block;
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  const int c8 = blockIdx.y * cg_per_block + cg_local;
  if (c8 * 8 >= C) return;
  const int C8 = C / 8;
  float gi[8], t2[8], c0[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = c8 * 8 + j;
    gi[j] = par[c];
    t2[j] = par[C + c] * inv_n;
    c0[j] = par[2 * C + c] * inv_n;
  }
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
#pragma unroll 2
  for (long r = row0; r < rows; r += step) {
    const long i = r * C8 + c8;
    s16x8 gv = ((const s16x8*)dy)[i];
    s16x8 xv = ((const s16x8*)x)[i];
    unsigned char mb = mask[i];
    s16x8 odx;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bits2f(gv[j]);
      g = (mb >> j) & 1 ? g : 0.f;
      odx[j] = f2bits(gi[j] * g - t2[j] * bits2f(xv[j]) + c0[j]);
    }
    ((s16x8*)dx)[i] = odx;
  }
}

// ---- v1: unroll 4 --------------------------------------------------------
__global__ void bwd_v1(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                       const unsigned char* __restrict__ mask,
                       const float* __restrict__ par, float inv_n,
                       bf16* __restrict__ dx, long rows, int C, int cg_per_block) {
  const int cg_local = threadIdx.x % cg_per_block;
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  const int c8 = blockIdx.y * cg_per_block + cg_local;
  if (c8 * 8 >= C) return;
  const int C8 = C / 8;
  float gi[8], t2[8], c0[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = c8 * 8 + j;
    gi[j] = par[c];
    t2[j] = par[C + c] * inv_n;
    c0[j] = par[2 * C + c] * inv_n;
  }
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
#pragma unroll 4
  for (long r = row0; r < rows; r += step) {
    const long i = r * C8 + c8;
    s16x8 gv = ((const s16x8*)dy)[i];
    s16x8 xv = ((const s16x8*)x)[i];
    unsigned char mb = mask[i];
    s16x8 odx;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bits2f(gv[j]);
      g = (mb >> j) & 1 ? g : 0.f;
      odx[j] = f2bits(gi[j] * g - t2[j] * bits2f(xv[j]) + c0[j]);
    }
    ((s16x8*)dx)[i] = odx;
  }
}

// ---- v2: manual 4-deep pipeline (all loads issued before any compute) ----
__global__ void bwd_v2(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                       const unsigned char* __restrict__ mask,
                       const float* __restrict__ par, float inv_n,
                       bf16* __restrict__ dx, long rows, int C, int cg_per_block) {
  const int cg_local = threadIdx.x % cg_per_block;
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  const int c8 = blockIdx.y * cg_per_block + cg_local;
  if (c8 * 8 >= C) return;
  const int C8 = C / 8;
  float gi[8], t2[8], c0[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = c8 * 8 + j;
    gi[j] = par[c];
    t2[j] = par[C + c] * inv_n;
    c0[j] = par[2 * C + c] * inv_n;
  }
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
  long r = row0;
  for (; r + 3 * step < rows; r += 4 * step) {
    s16x8 gv[4], xv[4];
    unsigned char mb[4];
    long idx[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      idx[u] = (r + u * step) * C8 + c8;
      gv[u] = ((const s16x8*)dy)[idx[u]];
      xv[u] = ((const s16x8*)x)[idx[u]];
      mb[u] = mask[idx[u]];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      s16x8 odx;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = bits2f(gv[u][j]);
        g = (mb[u] >> j) & 1 ? g : 0.f;
        odx[j] = f2bits(gi[j] * g - t2[j] * bits2f(xv[u][j]) + c0[j]);
      }
      ((s16x8*)dx)[idx[u]] = odx;
    }
  }
  for (; r < rows; r += step) {
    const long i = r * C8 + c8;
    s16x8 gv = ((const s16x8*)dy)[i];
    s16x8 xv = ((const s16x8*)x)[i];
    unsigned char mb = mask[i];
    s16x8 odx;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bits2f(gv[j]);
      g = (mb >> j) & 1 ? g : 0.f;
      odx[j] = f2bits(gi[j] * g - t2[j] * bits2f(xv[j]) + c0[j]);
    }
    ((s16x8*)dx)[i] = odx;
  }
}

// ---- v3: v0 + nontemporal loads/stores (bypass L2 allocation) ------------
__global__ void bwd_v3(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                       const unsigned char* __restrict__ mask,
                       const float* __restrict__ par, float inv_n,
                       bf16* __restrict__ dx, long rows, int C, int cg_per_block) {
  const int cg_local = threadIdx.x % cg_per_block;
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  const int c8 = blockIdx.y * cg_per_block + cg_local;
  if (c8 * 8 >= C) return;
  const int C8 = C / 8;
  float gi[8], t2[8], c0[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = c8 * 8 + j;
    gi[j] = par[c];
    t2[j] = par[C + c] * inv_n;
    c0[j] = par[2 * C + c] * inv_n;
  }
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
#pragma unroll 2
  for (long r = row0; r < rows; r += step) {
    const long i = r * C8 + c8;
    s16x8 gv = __builtin_nontemporal_load((const s16x8*)dy + i);
    s16x8 xv = __builtin_nontemporal_load((const s16x8*)x + i);
    unsigned char mb = mask[i];
    s16x8 odx;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bits2f(gv[j]);
      g = (mb >> j) & 1 ? g : 0.f;
      odx[j] = f2bits(gi[j] * g - t2[j] * bits2f(xv[j]) + c0[j]);
    }
    __builtin_nontemporal_store(odx, (s16x8*)dx + i);
  }
}

// ---- v4: 32 B per thread (two adjacent c8 chunks), 2-deep pipeline -------
// lanes cover C8/2 chunk-pairs; needs C8 even.
__global__ void bwd_v4(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                       const unsigned char* __restrict__ mask,
                       const float* __restrict__ par, float inv_n,
                       bf16* __restrict__ dx, long rows, int C, int cg_per_block) {
  const int cg_local = threadIdx.x % cg_per_block;   // c16 group now
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  const int c16 = blockIdx.y * cg_per_block + cg_local;
  if (c16 * 16 >= C) return;
  const int C8 = C / 8;
  float gi[16], t2[16], c0[16];
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int c = c16 * 16 + j;
    gi[j] = par[c];
    t2[j] = par[C + c] * inv_n;
    c0[j] = par[2 * C + c] * inv_n;
  }
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
#pragma unroll 2
  for (long r = row0; r < rows; r += step) {
    const long i = r * C8 + c16 * 2;
    s16x8 gv0 = ((const s16x8*)dy)[i], gv1 = ((const s16x8*)dy)[i + 1];
    s16x8 xv0 = ((const s16x8*)x)[i], xv1 = ((const s16x8*)x)[i + 1];
    unsigned char mb0 = mask[i], mb1 = mask[i + 1];
    s16x8 o0, o1;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bits2f(gv0[j]);
      g = (mb0 >> j) & 1 ? g : 0.f;
      o0[j] = f2bits(gi[j] * g - t2[j] * bits2f(xv0[j]) + c0[j]);
      float h = bits2f(gv1[j]);
      h = (mb1 >> j) & 1 ? h : 0.f;
      o1[j] = f2bits(gi[8 + j] * h - t2[8 + j] * bits2f(xv1[j]) + c0[8 + j]);
    }
    ((s16x8*)dx)[i] = o0;
    ((s16x8*)dx)[i + 1] = o1;
  }
}

// ---- roofline: 2-read 1-write triad, dwordx4, no math --------------------
__global__ void triad_kernel(const f32x4* __restrict__ a, const f32x4* __restrict__ b,
                             f32x4* __restrict__ c, long n4) {
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long step = (long)gridDim.x * blockDim.x;
#pragma unroll 4
  for (long i = i0; i < n4; i += step) {
    f32x4 va = a[i], vb = b[i];
    c[i] = va + vb;
  }
}

static inline int cgpb(int C) {
  int c8 = C / 8, cg = 1;
  while (cg < 32 && cg * 2 <= c8 && (c8 % (cg * 2)) == 0) cg *= 2;
  return cg;
}

extern "C" void run_bwd(int variant, const void* dy, const void* x, const void* mask,
                        const float* par, float inv_n, void* dx, long rows, int C,
                        int max_blocks) {
  int cg = cgpb(C);
  int unit = 8;
  if (variant == 4) {  // 16-channel unit
    int c16 = C / 16;
    cg = 1;
    while (cg < 32 && cg * 2 <= c16 && (c16 % (cg * 2)) == 0) cg *= 2;
    unit = 16;
  }
  const int rpb = 256 / cg;
  int row_blocks = (int)min((rows + rpb - 1) / rpb, (long)max_blocks);
  dim3 grid(row_blocks, (C / unit + cg - 1) / cg), block(256);
  switch (variant) {
    case 0: hipLaunchKernelGGL(bwd_v0, grid, block, 0, 0, (const bf16*)dy, (const bf16*)x, (const unsigned char*)mask, par, inv_n, (bf16*)dx, rows, C, cg); break;
    case 1: hipLaunchKernelGGL(bwd_v1, grid, block, 0, 0, (const bf16*)dy, (const bf16*)x, (const unsigned char*)mask, par, inv_n, (bf16*)dx, rows, C, cg); break;
    case 2: hipLaunchKernelGGL(bwd_v2, grid, block, 0, 0, (const bf16*)dy, (const bf16*)x, (const unsigned char*)mask, par, inv_n, (bf16*)dx, rows, C, cg); break;
    case 3: hipLaunchKernelGGL(bwd_v3, grid, block, 0, 0, (const bf16*)dy, (const bf16*)x, (const unsigned char*)mask, par, inv_n, (bf16*)dx, rows, C, cg); break;
    case 4: hipLaunchKernelGGL(bwd_v4, grid, block, 0, 0, (const bf16*)dy, (const bf16*)x, (const unsigned char*)mask, par, inv_n, (bf16*)dx, rows, C, cg); break;
  }
}

extern "C" void run_triad(const void* a, const void* b, void* c, long n4, int blocks) {
  hipLaunchKernelGGL(triad_kernel, dim3(blocks), dim3(256), 0, 0, (const f32x4*)a,
                     (const f32x4*)b, (f32x4*)c, n4);
}

extern "C" void dev_sync() { hipDeviceSynchronize(); }
