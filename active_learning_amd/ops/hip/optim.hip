// Fused optimizer updates (fp32 master params).
// SGD: p -= lr * (momentum-buffered (g + wd*p)); one read-modify-write pass
// (reference reaches torch.optim.SGD, arg_pools/default.py:39-40).
// Adam for VAAL's VAE/discriminator (vaal_sampler.py:139-140).

#include "al_common.h"

template <bool SHADOW>
__global__ void sgd_kernel(float* __restrict__ p, const float* __restrict__ g,
                           float* __restrict__ buf, float lr, float momentum,
                           float wd, long n, bf16* __restrict__ shadow) {
  for (long i = grid_stride_begin(); i < n; i += grid_stride_step()) {
    float grad = g[i] + wd * p[i];
    if (momentum != 0.f) {
      const float b = buf[i] * momentum + grad;
      buf[i] = b;
      grad = b;
    }
    const float np = p[i] - lr * grad;
    p[i] = np;
    if (SHADOW) shadow[i] = f2bf(np);  // refresh the bf16 weight copy in-pass
  }
}

extern "C" void al_sgd_step(float* p, const float* g, float* buf, float lr,
                            float momentum, float wd, long n, void* shadow,
                            hipStream_t stream) {
  int blocks = (int)min((n + 255) / 256, (long)2048);
  if (shadow)
    hipLaunchKernelGGL((sgd_kernel<true>), dim3(blocks), dim3(256), 0, stream, p, g,
                       buf, lr, momentum, wd, n, (bf16*)shadow);
  else
    hipLaunchKernelGGL((sgd_kernel<false>), dim3(blocks), dim3(256), 0, stream, p, g,
                       buf, lr, momentum, wd, n, nullptr);
}

// ---------------------------------------------------------------------------
// Multi-tensor SGD: ONE launch updates every parameter (the per-tensor
// variant cost 161 launches/step on ResNet-50, profiles/
// r50_b256_step_breakdown_final.md). The host packs a chunk table
// (6 int64 per chunk: p, g, momentum-buf, bf16-shadow pointers + element
// offset + count); each workgroup owns one chunk. Scalar fp32 accesses:
// grads may be element-aligned views into a DDP bucket, so 16-byte vector
// loads are not safe, and the pass is HBM-bound anyway.
struct SgdChunk {
  float* p;
  const float* g;
  float* buf;
  bf16* shadow;
  long off;
  long n;
};

// ZG (persistent-grad mode): zero the gradient in the same pass after
// consuming it — the training step then needs NO zero_grad / grad realloc,
// which keeps gradient storage stable so a hipGraph-captured step contains
// no allocations and the chunk table never rebuilds inside capture.
template <bool ZG>
__global__ void sgd_mt_kernel(const SgdChunk* __restrict__ table, float lr,
                              float momentum, float wd) {
  SgdChunk e = table[blockIdx.x];
  float* p = e.p + e.off;
  float* g = const_cast<float*>(e.g) + e.off;
  float* buf = e.buf ? e.buf + e.off : nullptr;
  bf16* sh = e.shadow ? e.shadow + e.off : nullptr;
  const int n = (int)e.n;
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    float grad = g[i] + wd * p[i];
    if (ZG) g[i] = 0.f;
    if (momentum != 0.f) {
      const float b = buf[i] * momentum + grad;
      buf[i] = b;
      grad = b;
    }
    const float np = p[i] - lr * grad;
    p[i] = np;
    if (sh) sh[i] = f2bf(np);
  }
}

extern "C" void al_sgd_step_multi(const void* table, int nchunks, float lr,
                                  float momentum, float wd, int zero_grad,
                                  hipStream_t stream) {
  if (zero_grad)
    hipLaunchKernelGGL((sgd_mt_kernel<true>), dim3(nchunks), dim3(256), 0, stream,
                       (const SgdChunk*)table, lr, momentum, wd);
  else
    hipLaunchKernelGGL((sgd_mt_kernel<false>), dim3(nchunks), dim3(256), 0, stream,
                       (const SgdChunk*)table, lr, momentum, wd);
}

// Variant reading (lr, momentum, wd) from device memory: lets a hipGraph-
// captured training step track the LR schedule — the host updates the
// 3-float hyper buffer between replays instead of re-capturing.
template <bool ZG>
__global__ void sgd_mt_kernel_dev(const SgdChunk* __restrict__ table,
                                  const float* __restrict__ hyper) {
  const float lr = hyper[0], momentum = hyper[1], wd = hyper[2];
  SgdChunk e = table[blockIdx.x];
  float* p = e.p + e.off;
  float* g = const_cast<float*>(e.g) + e.off;
  float* buf = e.buf ? e.buf + e.off : nullptr;
  bf16* sh = e.shadow ? e.shadow + e.off : nullptr;
  const int n = (int)e.n;
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    float grad = g[i] + wd * p[i];
    if (ZG) g[i] = 0.f;
    if (momentum != 0.f) {
      const float b = buf[i] * momentum + grad;
      buf[i] = b;
      grad = b;
    }
    const float np = p[i] - lr * grad;
    p[i] = np;
    if (sh) sh[i] = f2bf(np);
  }
}

extern "C" void al_sgd_step_multi_dev(const void* table, int nchunks,
                                      const float* hyper, int zero_grad,
                                      hipStream_t stream) {
  if (zero_grad)
    hipLaunchKernelGGL((sgd_mt_kernel_dev<true>), dim3(nchunks), dim3(256), 0,
                       stream, (const SgdChunk*)table, hyper);
  else
    hipLaunchKernelGGL((sgd_mt_kernel_dev<false>), dim3(nchunks), dim3(256), 0,
                       stream, (const SgdChunk*)table, hyper);
}

// ---------------------------------------------------------------------------
// Batched (K,R,S,C) -> (C,R,S,K) bf16 weight transpose: refreshes every
// conv's cached bwd-data permutation in ONE launch right after the fused SGD
// update (the per-conv ATen permute+clone path cost ~53 launches / 0.35 ms
// per step at B=256, tools/fill_audit.py). One workgroup per 64x64 (k, c)
// tile of one rs plane; host enqueues only full tiles (K, C % 64 == 0 —
// the igemm-eligible conv weights). Table row: 4 int64 =
// (src, dst, k0 | c0<<16 | rs<<32, K | C<<16 | RS<<32).
__global__ void wt_refresh_kernel(const long* __restrict__ table) {
  const long* row = table + (long)blockIdx.x * 4;
  const bf16* __restrict__ src = (const bf16*)row[0];
  bf16* __restrict__ dst = (bf16*)row[1];
  const long m1 = row[2], m2 = row[3];
  const int k0 = (int)(m1 & 0xffff), c0 = (int)((m1 >> 16) & 0xffff);
  const int rs = (int)(m1 >> 32);
  const int K = (int)(m2 & 0xffff), C = (int)((m2 >> 16) & 0xffff);
  const int RS = (int)(m2 >> 32);
  __shared__ short tile[64][68];  // 68: de-phase LDS banks across rows
  // load: thread -> (k row = t/4, 16-col strip = (t%4)*16); 2 x 16B per lane
  const int kl = threadIdx.x >> 2, cs = (threadIdx.x & 3) << 4;
  {
    const s16x8* sp =
        (const s16x8*)(src + ((long)(k0 + kl) * RS + rs) * C + c0 + cs);
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      s16x8 v = sp[u];
#pragma unroll
      for (int j = 0; j < 8; ++j) tile[kl][cs + u * 8 + j] = v[j];
    }
  }
  __syncthreads();
  // store transposed: thread -> (c row = t/4, 16-k strip)
  const int cl = threadIdx.x >> 2, ks = (threadIdx.x & 3) << 4;
  s16x8* dp = (s16x8*)(dst + ((long)(c0 + cl) * RS + rs) * K + k0 + ks);
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    s16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = tile[ks + u * 8 + j][cl];
    dp[u] = v;
  }
}

extern "C" void al_wt_refresh(const void* table, int nrows, hipStream_t stream) {
  hipLaunchKernelGGL(wt_refresh_kernel, dim3(nrows), dim3(256), 0, stream,
                     (const long*)table);
}

__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v, float lr,
                            float b1, float b2, float eps, float wd, float bc1,
                            float bc2, long n) {
  for (long i = grid_stride_begin(); i < n; i += grid_stride_step()) {
    const float grad = g[i] + wd * p[i];
    const float mi = m[i] * b1 + (1.f - b1) * grad;
    const float vi = v[i] * b2 + (1.f - b2) * grad * grad;
    m[i] = mi;
    v[i] = vi;
    p[i] -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
  }
}

extern "C" void al_adam_step(float* p, const float* g, float* m, float* v, float lr,
                             float b1, float b2, float eps, float wd, float bc1,
                             float bc2, long n, hipStream_t stream) {
  int blocks = (int)min((n + 255) / 256, (long)2048);
  hipLaunchKernelGGL(adam_kernel, dim3(blocks), dim3(256), 0, stream, p, g, m, v, lr,
                     b1, b2, eps, wd, bc1, bc2, n);
}
