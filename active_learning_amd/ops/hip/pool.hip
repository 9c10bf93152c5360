// NHWC pooling kernels (ResNet stem maxpool 3x3 s2 + global average pool).
// Reference counterpart: torchvision resnet maxpool/avgpool.

#include "al_common.h"

// ---------------------------------------------------------------------------
// maxpool fwd: y[n,p,q,c] = max window; idx = flat h*w argmax (torch format,
// so the CPU fallback and the GPU kernel share the backward contract)
// ---------------------------------------------------------------------------

__global__ void maxpool_fwd_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                                   int* __restrict__ idx, int N, int H, int W, int C,
                                   int P, int Q, int kernel, int stride, int pad) {
  const long total = (long)N * P * Q * C;
  for (long i = grid_stride_begin(); i < total; i += grid_stride_step()) {
    const int c = (int)(i % C);
    long t = i / C;
    const int q = (int)(t % Q);
    t /= Q;
    const int p = (int)(t % P);
    const int n = (int)(t / P);
    float best = -3.0e38f;
    int best_hw = 0;
    const int h0 = p * stride - pad, w0 = q * stride - pad;
    for (int r = 0; r < kernel; ++r) {
      const int h = h0 + r;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < kernel; ++s) {
        const int w = w0 + s;
        if (w < 0 || w >= W) continue;
        const float v = bf2f(x[(((long)n * H + h) * W + w) * C + c]);
        if (v > best) { best = v; best_hw = h * W + w; }
      }
    }
    y[i] = f2bf(best);
    idx[i] = best_hw;
  }
}

// vectorized variants (C % 8 == 0, the stem's C=64): one thread owns an
// 8-channel chunk, so window scans move s16x8/int4 lines instead of scalars
// and the (n,p,q) decode happens once per 8 channels.

typedef int i32x4 __attribute__((ext_vector_type(4)));

__global__ void maxpool_fwd8_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                                    int* __restrict__ idx, int N, int H, int W,
                                    int C8, int P, int Q, int kernel, int stride,
                                    int pad) {
  const long total = (long)N * P * Q * C8;
  for (long i = grid_stride_begin(); i < total; i += grid_stride_step()) {
    const int c8 = (int)(i % C8);
    long t = i / C8;
    const int q = (int)(t % Q);
    t /= Q;
    const int p = (int)(t % P);
    const int n = (int)(t / P);
    float best[8];
    int bhw[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { best[j] = -3.0e38f; bhw[j] = 0; }
    const int h0 = p * stride - pad, w0 = q * stride - pad;
    for (int r = 0; r < kernel; ++r) {
      const int h = h0 + r;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < kernel; ++s) {
        const int w = w0 + s;
        if (w < 0 || w >= W) continue;
        const s16x8 v = ((const s16x8*)x)[((long)n * H + h) * W * C8 + w * C8 + c8];
        const int hw = h * W + w;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = bits2f(v[j]);
          if (f > best[j]) { best[j] = f; bhw[j] = hw; }
        }
      }
    }
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bits(best[j]);
    ((s16x8*)y)[i] = o;
    i32x4 o0 = {bhw[0], bhw[1], bhw[2], bhw[3]};
    i32x4 o1 = {bhw[4], bhw[5], bhw[6], bhw[7]};
    ((i32x4*)idx)[i * 2] = o0;
    ((i32x4*)idx)[i * 2 + 1] = o1;
  }
}

__global__ void maxpool_bwd8_kernel(const bf16* __restrict__ dy,
                                    const int* __restrict__ idx,
                                    bf16* __restrict__ dx, int N, int H, int W,
                                    int C8, int P, int Q, int kernel, int stride,
                                    int pad) {
  const long total = (long)N * H * W * C8;
  for (long i = grid_stride_begin(); i < total; i += grid_stride_step()) {
    const int c8 = (int)(i % C8);
    long t = i / C8;
    const int w = (int)(t % W);
    t /= W;
    const int h = (int)(t % H);
    const int n = (int)(t / H);
    const int my_hw = h * W + w;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    const int pmin = max(0, (h + pad - kernel + stride) / stride);
    const int pmax = min(P - 1, (h + pad) / stride);
    const int qmin = max(0, (w + pad - kernel + stride) / stride);
    const int qmax = min(Q - 1, (w + pad) / stride);
    for (int p = pmin; p <= pmax; ++p)
      for (int q = qmin; q <= qmax; ++q) {
        const long o = ((long)n * P + p) * Q * C8 + q * C8 + c8;
        const i32x4 i0 = ((const i32x4*)idx)[o * 2];
        const i32x4 i1 = ((const i32x4*)idx)[o * 2 + 1];
        const s16x8 g = ((const s16x8*)dy)[o];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          if (i0[j] == my_hw) acc[j] += bits2f(g[j]);
          if (i1[j] == my_hw) acc[4 + j] += bits2f(g[4 + j]);
        }
      }
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) o8[j] = f2bits(acc[j]);
    ((s16x8*)dx)[i] = o8;
  }
}

extern "C" void al_maxpool_fwd(const void* x, void* y, int* idx, int N, int H, int W,
                               int C, int P, int Q, int kernel, int stride, int pad,
                               hipStream_t stream) {
  if (C % 8 == 0) {
    long total = (long)N * P * Q * (C / 8);
    int blocks = (int)min((total + 255) / 256, (long)4096);
    hipLaunchKernelGGL(maxpool_fwd8_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const bf16*)x, (bf16*)y, idx, N, H, W, C / 8, P, Q, kernel,
                       stride, pad);
    return;
  }
  long total = (long)N * P * Q * C;
  int blocks = (int)min((total + 255) / 256, (long)4096);
  hipLaunchKernelGGL(maxpool_fwd_kernel, dim3(blocks), dim3(256), 0, stream,
                     (const bf16*)x, (bf16*)y, idx, N, H, W, C, P, Q, kernel, stride,
                     pad);
}

// backward: per INPUT pixel, sum dy over covering windows whose argmax == me
// (deterministic, no atomics; each input is covered by <= ceil(k/s)^2 windows)
__global__ void maxpool_bwd_kernel(const bf16* __restrict__ dy,
                                   const int* __restrict__ idx,
                                   bf16* __restrict__ dx, int N, int H, int W, int C,
                                   int P, int Q, int kernel, int stride, int pad) {
  const long total = (long)N * H * W * C;
  for (long i = grid_stride_begin(); i < total; i += grid_stride_step()) {
    const int c = (int)(i % C);
    long t = i / C;
    const int w = (int)(t % W);
    t /= W;
    const int h = (int)(t % H);
    const int n = (int)(t / H);
    const int my_hw = h * W + w;
    float acc = 0.f;
    const int pmin = max(0, (h + pad - kernel + stride) / stride);
    const int pmax = min(P - 1, (h + pad) / stride);
    const int qmin = max(0, (w + pad - kernel + stride) / stride);
    const int qmax = min(Q - 1, (w + pad) / stride);
    for (int p = pmin; p <= pmax; ++p)
      for (int q = qmin; q <= qmax; ++q) {
        const long o = (((long)n * P + p) * Q + q) * C + c;
        if (idx[o] == my_hw) acc += bf2f(dy[o]);
      }
    dx[i] = f2bf(acc);
  }
}

extern "C" void al_maxpool_bwd(const void* dy, const int* idx, void* dx, int N, int H,
                               int W, int C, int P, int Q, int kernel, int stride,
                               int pad, hipStream_t stream) {
  if (C % 8 == 0) {
    long total = (long)N * H * W * (C / 8);
    int blocks = (int)min((total + 255) / 256, (long)4096);
    hipLaunchKernelGGL(maxpool_bwd8_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const bf16*)dy, idx, (bf16*)dx, N, H, W, C / 8, P, Q, kernel,
                       stride, pad);
    return;
  }
  long total = (long)N * H * W * C;
  int blocks = (int)min((total + 255) / 256, (long)4096);
  hipLaunchKernelGGL(maxpool_bwd_kernel, dim3(blocks), dim3(256), 0, stream,
                     (const bf16*)dy, idx, (bf16*)dx, N, H, W, C, P, Q, kernel, stride,
                     pad);
}

// ---------------------------------------------------------------------------
// global average pool: (N,H,W,C) -> (N,C) fp32-accumulated, bf16-out
// block per image; lanes stride channels; rows streamed
// ---------------------------------------------------------------------------

__global__ void gap_kernel(const bf16* __restrict__ x, bf16* __restrict__ y, int HW,
                           int C, int cg_per_block) {
  // thread -> (c8 chunk, row lane): 16B vector loads, coalesced across lanes
  const int n = blockIdx.x;
  const int C8 = C / 8;
  const int cg = blockIdx.y * cg_per_block + (threadIdx.x % cg_per_block);
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  if (cg >= C8) return;
  const float inv = 1.0f / HW;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const s16x8* base = (const s16x8*)x + (long)n * HW * C8 + cg;
  for (int r = row_lane; r < HW; r += rows_per_block) {
    s16x8 v = base[(long)r * C8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += bits2f(v[j]);
  }
  __shared__ float red[256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x][j] = acc[j];
  __syncthreads();
  if (row_lane == 0) {
    for (int rl = 1; rl < rows_per_block; ++rl)
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += red[rl * cg_per_block + (threadIdx.x % cg_per_block)][j];
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bits(acc[j] * inv);
    ((s16x8*)y)[(long)n * C8 + cg] = o;
  }
}

extern "C" void al_global_avg_pool(const void* x, void* y, int N, int HW, int C,
                                   hipStream_t stream) {
  int c8 = C / 8;
  int cg = 1;
  while (cg < 32 && cg * 2 <= c8 && (c8 % (cg * 2)) == 0) cg *= 2;
  dim3 grid(N, (c8 + cg - 1) / cg);
  hipLaunchKernelGGL(gap_kernel, grid, dim3(256), 0, stream, (const bf16*)x,
                     (bf16*)y, HW, C, cg);
}
