// Common device helpers for the MI355X (gfx950 / CDNA4) kernels.
// Wavefront = 64 lanes; LDS 160 KiB/CU; 8 XCDs x 32 CUs.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define AL_DEV __device__ __forceinline__

using bf16 = __hip_bfloat16;

// vector types for wide loads (G13: always vectorize bf16 as short4/short8)
typedef short  s16x8 __attribute__((ext_vector_type(8)));
typedef short  s16x4 __attribute__((ext_vector_type(4)));
typedef float  f32x4 __attribute__((ext_vector_type(4)));
typedef float  f32x2 __attribute__((ext_vector_type(2)));
typedef short  bf16x8_raw __attribute__((ext_vector_type(8)));

AL_DEV float bf2f(bf16 v) { return __bfloat162float(v); }
AL_DEV bf16 f2bf(float v) { return __float2bfloat16(v); }

AL_DEV float bits2f(short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}
AL_DEV short f2bits(float f) {
  union { unsigned int i; float f; } c;
  c.f = f;
  unsigned int x = c.i;
  // round-to-nearest-even bf16
  unsigned int lsb = (x >> 16) & 1u;
  x += 0x7fffu + lsb;
  return (short)(x >> 16);
}

// full-wave (64-lane) reductions
AL_DEV float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}
AL_DEV float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}
AL_DEV float wave_bcast(float v, int lane) { return __shfl(v, lane, 64); }

constexpr int kWave = 64;

AL_DEV long grid_stride_begin() {
  return (long)blockIdx.x * blockDim.x + threadIdx.x;
}
AL_DEV long grid_stride_step() { return (long)gridDim.x * blockDim.x; }

static inline int ceil_div_host(long a, long b) { return (int)((a + b - 1) / b); }
