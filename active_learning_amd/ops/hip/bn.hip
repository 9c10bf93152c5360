// Fused BatchNorm(+residual+ReLU) over NHWC bf16, CDNA4.
//
// Replaces the cuDNN BN+ReLU kernels the reference reaches implicitly
// (frozen-stats eval variant = reference strategy.py:366-367).
// Split-kernel design so SyncBN's cross-rank all-reduce (RCCL) slots between
// the stats pass and the normalize pass:
//   bn_stats       : per-channel sum / sum-of-squares (fp32)
//   bn_norm_fwd    : y = act((x - mean) * invstd * g + b [+ residual])
//   bn_bwd_reduce  : per-channel sums of dy~ and dy~*xhat (dy~ = mask * dy)
//   bn_bwd         : dx (+dres) from the reduced terms
// All element passes are memory-bound: bf16x8 (16 B/lane) vectorized, one
// read of each operand, grid-stride (G11/G13).

#include <cstdlib>

#include "al_common.h"

// ---------------------------------------------------------------------------
// stats: rows = N*H*W, channels C (C % 8 == 0 for the vector path)
// one block covers a row-slab for a 64-channel group; partial sums -> atomics
// ---------------------------------------------------------------------------

__global__ void bn_stats_kernel(const bf16* __restrict__ x, float* __restrict__ sum,
                                float* __restrict__ sumsq, long rows, int C,
                                int cg_per_block) {
  // thread -> (c8 group, row lane); consecutive lanes read consecutive 16B
  // chunks (s16x8), so a 32-lane group streams 512 contiguous bytes.
  const int cg_local = threadIdx.x % cg_per_block;
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  const int c8 = blockIdx.y * cg_per_block + cg_local;
  if (c8 * 8 >= C) return;
  const int C8 = C / 8;
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0}, ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
  for (long r = row0; r < rows; r += step) {
    s16x8 v = ((const s16x8*)x)[r * C8 + c8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = bits2f(v[j]);
      s[j] += f;
      ss[j] += f * f;
    }
  }
  // reduce across row lanes sharing a c8 group through LDS
  __shared__ float red[2][256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][threadIdx.x][j] = s[j];
    red[1][threadIdx.x][j] = ss[j];
  }
  __syncthreads();
  if (row_lane == 0) {
    for (int rl = 1; rl < rows_per_block; ++rl)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        s[j] += red[0][rl * cg_per_block + cg_local][j];
        ss[j] += red[1][rl * cg_per_block + cg_local][j];
      }
    // per-block partials: sum[2][gridDim.x][C]; the caller reduces over
    // blocks (atomics on few addresses serialize at this block count)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sum[(long)blockIdx.x * C + c8 * 8 + j] = s[j];
      sumsq[(long)blockIdx.x * C + c8 * 8 + j] = ss[j];
    }
  }
}

// Nontemporal load/store for the element passes when the streams are far
// bigger than L2 (32 MiB across the 8 XCDs): measured +10-12% on the
// >=64 MiB/stream shapes and -17% on the L2-reachable 7x7x2048 shape
// (tools/bn_ab.py), so the launchers pick per launch by footprint.
template <bool NT>
AL_DEV s16x8 ld8(const s16x8* p) {
  if constexpr (NT) return __builtin_nontemporal_load(p);
  else return *p;
}
template <bool NT>
AL_DEV void st8(s16x8* p, s16x8 v) {
  if constexpr (NT) __builtin_nontemporal_store(v, p);
  else *p = v;
}

static inline bool bn_nt(long rows, int C) {
  static const bool on = [] {
    const char* e = getenv("AL_BN_NT");
    return !(e && e[0] == '0');
  }();
  return on && rows * (long)C * 2 >= (64L << 20);
}

static inline int bn_cg_per_block(int C) {
  int c8 = C / 8;
  int cg = 1;
  while (cg < 32 && cg * 2 <= c8 && (c8 % (cg * 2)) == 0) cg *= 2;
  return cg;  // largest power-of-two divisor of C/8, capped at 32
}

static inline int bn_row_block_cap(int C) {
  // bound the partial buffer (nb x C fp32) to ~512 KiB: wide layers need
  // fewer row blocks, narrow layers keep the full 2048 for bandwidth
  long cap = 131072 / C;
  if (cap < 128) cap = 128;
  if (cap > 2048) cap = 2048;
  return (int)cap;
}

extern "C" void al_bn_stats(const void* x, float* sum, float* sumsq, long rows, int C,
                            hipStream_t stream) {
  dim3 block(256);
  const int cg = bn_cg_per_block(C);
  const int rpb = 256 / cg;
  int row_blocks = (int)min((rows + rpb - 1) / rpb, (long)bn_row_block_cap(C));
  dim3 grid(row_blocks, (C / 8 + cg - 1) / cg);
  hipLaunchKernelGGL(bn_stats_kernel, grid, block, 0, stream, (const bf16*)x, sum,
                     sumsq, rows, C, cg);
}

extern "C" int al_bn_reduce_blocks(long rows, int C) {
  const int cg = bn_cg_per_block(C);
  const int rpb = 256 / cg;
  return (int)min((rows + rpb - 1) / rpb, (long)bn_row_block_cap(C));
}

// ---------------------------------------------------------------------------
// normalize forward (+ residual + relu), bf16x8 vectorized
// ---------------------------------------------------------------------------

// v2: fixed channel-group per thread (2D grid as bn_stats), so the four
// per-channel parameters fold into two registers (a, b) hoisted out of the
// row loop: the inner loop is load-16B / 8 fma / store-16B with no modulo
// and no parameter reloads (the v1 grid-stride form re-read mean/invstd/
// gamma/beta per chunk and spent a 64-bit modulo per 16B — measured ~3.2
// TB/s; this form matches the stats kernels' ~5+ TB/s streaming pattern).
template <bool RELU, bool RES, bool NT>
__global__ void bn_norm2_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                const bf16* __restrict__ res,
                                unsigned char* __restrict__ relu_mask,
                                long rows, int C, int cg_per_block) {
  const int cg_local = threadIdx.x % cg_per_block;
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  const int c8 = blockIdx.y * cg_per_block + cg_local;
  if (c8 * 8 >= C) return;
  const int C8 = C / 8;
  float a[8], b[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = c8 * 8 + j;
    a[j] = invstd[c] * gamma[c];
    b[j] = beta[c] - mean[c] * a[j];
  }
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
  #pragma unroll 2
  for (long r = row0; r < rows; r += step) {
    s16x8 v = ld8<NT>((const s16x8*)x + r * C8 + c8);
    s16x8 rv;
    if (RES) rv = ld8<NT>((const s16x8*)res + r * C8 + c8);
    s16x8 o;
    unsigned char mb = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bits2f(v[j]) * a[j] + b[j];
      if (RES) f += bits2f(rv[j]);
      if (RELU) {
        mb |= (unsigned char)(f > 0.f) << j;
        f = fmaxf(f, 0.f);
      }
      o[j] = f2bits(f);
    }
    st8<NT>((s16x8*)y + r * C8 + c8, o);
    // one mask BIT per element: the backward passes read this instead of
    // re-streaming y (drops a full activation read from both)
    if (RELU && relu_mask) relu_mask[r * C8 + c8] = mb;
  }
}

extern "C" void al_bn_norm_fwd(const void* x, void* y, const float* mean,
                               const float* invstd, const float* gamma,
                               const float* beta, const void* res, int relu,
                               void* relu_mask, long rows, int C,
                               hipStream_t stream) {
  const bf16* r = (const bf16*)res;
  const int cg = bn_cg_per_block(C);
  const int rpb = 256 / cg;
  int row_blocks = (int)min((rows + rpb - 1) / rpb, (long)4096);
  dim3 grid(row_blocks, (C / 8 + cg - 1) / cg), block(256);
#define CASE1(RELU_, RES_, NT_) \
  hipLaunchKernelGGL((bn_norm2_kernel<RELU_, RES_, NT_>), grid, block, 0, stream, \
                     (const bf16*)x, (bf16*)y, mean, invstd, gamma, beta, r, \
                     (unsigned char*)relu_mask, rows, C, cg)
#define CASE(RELU_, RES_) \
  do { if (bn_nt(rows, C)) CASE1(RELU_, RES_, true); else CASE1(RELU_, RES_, false); } while (0)
  if (relu) { if (r) CASE(true, true); else CASE(true, false); }
  else      { if (r) CASE(false, true); else CASE(false, false); }
#undef CASE
#undef CASE1
}

// ---------------------------------------------------------------------------
// backward reduce: sum_dy, sum_dy_xhat per channel (dy masked by y>0 if RELU)
// ---------------------------------------------------------------------------

template <bool RELU, bool NT>
__global__ void bn_bwd_reduce_kernel(const bf16* __restrict__ dy,
                                     const bf16* __restrict__ x,
                                     const unsigned char* __restrict__ relu_mask,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     float* __restrict__ sum_dy,
                                     float* __restrict__ sum_dy_xhat, long rows,
                                     int C, int cg_per_block) {
  const int cg_local = threadIdx.x % cg_per_block;
  const int row_lane = threadIdx.x / cg_per_block;
  const int rows_per_block = 256 / cg_per_block;
  const int c8 = blockIdx.y * cg_per_block + cg_local;
  if (c8 * 8 >= C) return;
  const int C8 = C / 8;
  float m[8], is[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    m[j] = mean[c8 * 8 + j];
    is[j] = invstd[c8 * 8 + j];
  }
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0}, sx[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
  for (long r = row0; r < rows; r += step) {
    const long off = r * C8 + c8;
    s16x8 gv = ld8<NT>((const s16x8*)dy + off);
    s16x8 xv = ld8<NT>((const s16x8*)x + off);
    unsigned char mb = 0xff;
    if (RELU) mb = relu_mask[off];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bits2f(gv[j]);
      if (RELU) g = (mb >> j) & 1 ? g : 0.f;
      s[j] += g;
      sx[j] += g * (bits2f(xv[j]) - m[j]) * is[j];
    }
  }
  __shared__ float red[2][256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][threadIdx.x][j] = s[j];
    red[1][threadIdx.x][j] = sx[j];
  }
  __syncthreads();
  if (row_lane == 0) {
    for (int rl = 1; rl < rows_per_block; ++rl)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        s[j] += red[0][rl * cg_per_block + cg_local][j];
        sx[j] += red[1][rl * cg_per_block + cg_local][j];
      }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sum_dy[(long)blockIdx.x * C + c8 * 8 + j] = s[j];
      sum_dy_xhat[(long)blockIdx.x * C + c8 * 8 + j] = sx[j];
    }
  }
}

extern "C" void al_bn_bwd_reduce(const void* dy, const void* x, const void* relu_mask,
                                 const float* mean, const float* invstd,
                                 float* sum_dy, float* sum_dy_xhat, int relu,
                                 long rows, int C, hipStream_t stream) {
  dim3 block(256);
  const int cg = bn_cg_per_block(C);
  const int rpb = 256 / cg;
  int row_blocks = (int)min((rows + rpb - 1) / rpb, (long)bn_row_block_cap(C));
  dim3 grid(row_blocks, (C / 8 + cg - 1) / cg);
#define CASE1(RELU_, NT_) \
  hipLaunchKernelGGL((bn_bwd_reduce_kernel<RELU_, NT_>), grid, block, 0, stream, \
                     (const bf16*)dy, (const bf16*)x, \
                     (const unsigned char*)relu_mask, mean, invstd, sum_dy, \
                     sum_dy_xhat, rows, C, cg)
#define CASE(RELU_) \
  do { if (bn_nt(rows, C)) CASE1(RELU_, true); else CASE1(RELU_, false); } while (0)
  if (relu) CASE(true);
  else CASE(false);
#undef CASE
#undef CASE1
}

// ---------------------------------------------------------------------------
// backward element pass:
//   batch stats: dx = g*istd * (dy~ - sum_dy/n - xhat * sum_dy_xhat/n)
//   frozen:      dx = g*istd * dy~
//   dres = dy~ when the forward fused a residual add
// ---------------------------------------------------------------------------

// v2 backward elementwise: same fixed-channel-strip structure; the batch
// terms fold to dx = gi*g - t2*x + c0 (two fma) with per-channel (gi, t2,
// c0) hoisted into registers.
template <bool RELU, bool BATCH, bool RES, bool NT>
__global__ void bn_bwd2_kernel(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                               const unsigned char* __restrict__ relu_mask,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const float* __restrict__ gamma,
                               const float* __restrict__ sum_dy,
                               const float* __restrict__ sum_dy_xhat, float inv_n,
                               bf16* __restrict__ dx, bf16* __restrict__ dres,
                               long rows, int C, int cg_per_block) {
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
    gi[j] = gamma[c] * invstd[c];
    if (BATCH) {
      t2[j] = gi[j] * (sum_dy_xhat[c] * inv_n) * invstd[c];
      c0[j] = t2[j] * mean[c] - gi[j] * (sum_dy[c] * inv_n);
    }
  }
  const long row0 = (long)blockIdx.x * rows_per_block + row_lane;
  const long step = (long)gridDim.x * rows_per_block;
  #pragma unroll 2
  for (long r = row0; r < rows; r += step) {
    const long i = r * C8 + c8;
    s16x8 gv = ld8<NT>((const s16x8*)dy + i);
    s16x8 xv;
    if (BATCH) xv = ld8<NT>((const s16x8*)x + i);
    unsigned char mb = 0xff;
    if (RELU) mb = relu_mask[i];
    s16x8 odx, ores;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bits2f(gv[j]);
      if (RELU) g = (mb >> j) & 1 ? g : 0.f;
      if (RES) ores[j] = f2bits(g);
      float v;
      if (BATCH) v = gi[j] * g - t2[j] * bits2f(xv[j]) + c0[j];
      else v = gi[j] * g;
      odx[j] = f2bits(v);
    }
    st8<NT>((s16x8*)dx + i, odx);
    if (RES) st8<NT>((s16x8*)dres + i, ores);
  }
}

extern "C" void al_bn_bwd(const void* dy, const void* x, const void* relu_mask,
                          const float* mean, const float* invstd, const float* gamma,
                          const float* sum_dy, const float* sum_dy_xhat, float n,
                          int use_batch_stats, int relu, int has_res, void* dx,
                          void* dres, long rows, int C, hipStream_t stream) {
  const int cg = bn_cg_per_block(C);
  const int rpb = 256 / cg;
  int row_blocks = (int)min((rows + rpb - 1) / rpb, (long)4096);
  dim3 grid(row_blocks, (C / 8 + cg - 1) / cg), block(256);
  float inv_n = 1.0f / n;
#define CASE1(RELU_, BATCH_, RES_, NT_) \
  hipLaunchKernelGGL((bn_bwd2_kernel<RELU_, BATCH_, RES_, NT_>), grid, block, 0, stream, \
                     (const bf16*)dy, (const bf16*)x, (const unsigned char*)relu_mask, mean, invstd, \
                     gamma, sum_dy, sum_dy_xhat, inv_n, (bf16*)dx, (bf16*)dres, \
                     rows, C, cg)
#define CASE(RELU_, BATCH_, RES_) \
  do { if (bn_nt(rows, C)) CASE1(RELU_, BATCH_, RES_, true); \
       else CASE1(RELU_, BATCH_, RES_, false); } while (0)
  if (relu) {
    if (use_batch_stats) { if (has_res) CASE(true, true, true); else CASE(true, true, false); }
    else                 { if (has_res) CASE(true, false, true); else CASE(true, false, false); }
  } else {
    if (use_batch_stats) { if (has_res) CASE(false, true, true); else CASE(false, true, false); }
    else                 { if (has_res) CASE(false, false, true); else CASE(false, false, false); }
  }
#undef CASE
#undef CASE1
}

// ---------------------------------------------------------------------------
// stage-2: reduce the per-block partials of BOTH arrays in one launch
// (replaces two torch .sum(0) calls per BN op — launch-bound at 53 BN layers)
// ---------------------------------------------------------------------------

__global__ void bn_part_reduce_kernel(const float* __restrict__ part_a,
                                      const float* __restrict__ part_b,
                                      float* __restrict__ out_a,
                                      float* __restrict__ out_b, int nb, int C) {
  // block = 8 channels x 32 block-lanes; each lane strides the nb partials,
  // then an LDS tree folds the 32 lanes per channel.
  const int cl = threadIdx.x & 7;            // channel within the block's 8
  const int bl = threadIdx.x >> 3;           // 32 b-lanes
  const int c = blockIdx.x * 8 + cl;
  __shared__ float red[2][256];
  float sa = 0.f, sb = 0.f;
  if (c < C) {
    for (int b = bl; b < nb; b += 32) {
      sa += part_a[(long)b * C + c];
      sb += part_b[(long)b * C + c];
    }
  }
  red[0][threadIdx.x] = sa;
  red[1][threadIdx.x] = sb;
  __syncthreads();
  for (int step = 128; step >= 8; step >>= 1) {
    if (threadIdx.x < step) {
      red[0][threadIdx.x] += red[0][threadIdx.x + step];
      red[1][threadIdx.x] += red[1][threadIdx.x + step];
    }
    __syncthreads();
  }
  if (threadIdx.x < 8 && c < C) {
    out_a[c] = red[0][threadIdx.x];
    out_b[c] = red[1][threadIdx.x];
  }
}

extern "C" void al_bn_part_reduce(const float* part_a, const float* part_b,
                                  float* out_a, float* out_b, int nb, int C,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(bn_part_reduce_kernel, dim3((C + 7) / 8), dim3(256), 0,
                     stream, part_a, part_b, out_a, out_b, nb, C);
}

// ---------------------------------------------------------------------------
// finalize: partials -> mean/invstd (+ in-place running-stat update), fusing
// what was ~6 tiny ATen ops per BN layer into the stage-2 reduce.
// ---------------------------------------------------------------------------

__global__ void bn_finalize_kernel(const float* __restrict__ part_s,
                                   const float* __restrict__ part_ss,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var, int nb, int C,
                                   float inv_n, float unbias, float momentum,
                                   float eps, int update_running) {
  const int cl = threadIdx.x & 7;
  const int bl = threadIdx.x >> 3;
  const int c = blockIdx.x * 8 + cl;
  __shared__ float red[2][256];
  float sa = 0.f, sb = 0.f;
  if (c < C) {
    for (int b = bl; b < nb; b += 32) {
      sa += part_s[(long)b * C + c];
      sb += part_ss[(long)b * C + c];
    }
  }
  red[0][threadIdx.x] = sa;
  red[1][threadIdx.x] = sb;
  __syncthreads();
  for (int step = 128; step >= 8; step >>= 1) {
    if (threadIdx.x < step) {
      red[0][threadIdx.x] += red[0][threadIdx.x + step];
      red[1][threadIdx.x] += red[1][threadIdx.x + step];
    }
    __syncthreads();
  }
  if (threadIdx.x < 8 && c < C) {
    const float m = red[0][threadIdx.x] * inv_n;
    float var = red[1][threadIdx.x] * inv_n - m * m;
    var = var < 0.f ? 0.f : var;
    mean[c] = m;
    invstd[c] = rsqrtf(var + eps);
    if (update_running) {
      running_mean[c] = running_mean[c] * (1.f - momentum) + m * momentum;
      running_var[c] = running_var[c] * (1.f - momentum) + var * unbias * momentum;
    }
  }
}

extern "C" void al_bn_finalize(const float* part_s, const float* part_ss, float* mean,
                               float* invstd, float* running_mean, float* running_var,
                               int nb, int C, float n, float momentum, float eps,
                               int update_running, hipStream_t stream) {
  const float unbias = n / (n - 1.f > 0.f ? n - 1.f : 1.f);
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 7) / 8), dim3(256), 0, stream,
                     part_s, part_ss, mean, invstd, running_mean, running_var, nb, C,
                     1.f / n, unbias, momentum, eps, update_running);
}
