// Implicit-GEMM convolution for NHWC bf16 on gfx950 (CDNA4 MFMA).
//
// Replaces the cuDNN conv kernels the reference reaches through
// torchvision/resnet forward+backward.
//
// Design (MI355X-first):
//  * forward:   y[M=N*P*Q, K] = im2col(x)[M, RSC] @ W^T      (W is (K,R,S,C))
//  * bwd-data:  dx[M=N*H*W, C] = scat(dY)[M, RSK] @ Wt       (Wt is (C,R,S,K),
//               k fastest in the contraction, pre-permuted host-side — weights
//               are tiny next to activations)
//    Both are the same GEMM with different per-row gather rules, so one
//    templated kernel (MODE) serves both; a transposed conv's forward IS the
//    bwd-data computation (models/layers.py).
//  * 128x128x64 tile, 4 waves (2x2), 64x64 per wave, mfma_f32_16x16x32_bf16,
//    fp32 accumulators.
//  * global->LDS staging via __builtin_amdgcn_global_load_lds width 16
//    (async, no VGPR round trip); LDS image is lane-linear so the
//    bank-conflict XOR swizzle (slot ^= row&7) is applied to the per-lane
//    SOURCE address and re-applied on the ds_read_b128 side (guide §5.4
//    rule 21: both-sides-or-neither).
//  * out-of-bounds / padding gathers read a device zero page: no branches in
//    the staging inner loop, no predication on the DMA.
//  * requires C % 8 == 0 and contraction-dim % 64 == 0 (all non-stem ResNet
//    and VAE shapes); other shapes take the direct fallback kernels below.
//
// Fragment layouts (verified by tests/test_kernels_gpu.py layout probe):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j], j=0..7
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C/D: lane l, reg r -> row = (l>>4)*4 + r, col = l&15

#include "al_common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#define MODE_FWD 0
#define MODE_BWD_DATA 1
#define MODE_BWD_S2 2    // stride-2 bwd-data, one (h%2, w%2) parity class per
                         // launch: contraction runs only the VALID (r,s) of
                         // the class, eliminating the 4x zero-chunk redundancy

struct ConvShape {
  int N, H, W, C;       // input (or dx) spatial shape
  int K;                // output channels (fwd) / dY channels (bwd-data)
  int R, S;             // filter
  int P, Q;             // output spatial (fwd) / dY spatial (bwd-data)
  int stride, pad;
  long M;               // GEMM rows: N*P*Q (fwd) or N*H*W (bwd)
  int Nout;             // GEMM cols: K (fwd) or C (bwd)
  int KD;               // contraction: R*S*C (fwd) or R*S*K (bwd)
  // MODE_BWD_S2 only (H/W hold the class spatial dims Hc/Wc; R/S hold the
  // class filter counts Rc/Sc):
  int Horig, Worig;     // full dx spatial dims
  int hcl, wcl;         // parity class
  int poff, qoff;       // p = (hh + poff) - ri, q = (ww + qoff) - si
  long b_rowstride;     // Rorig*Sorig*K (Wt row pitch)
  long b_rstride;       // 2*Sorig*K    (ri step within a Wt row)
  long b_coff;          // (rpar*Sorig + spar)*K (class base inside a row)
};

// ---------------------------------------------------------------------------
// per-mode gather: global source pointer for a 16-byte (8 bf16) chunk of the
// A matrix at (row m, kd..kd+7). Returns the zero page when out of bounds.
// ---------------------------------------------------------------------------

template <int MODE>
AL_DEV const bf16* a_chunk_ptr(const bf16* __restrict__ a, const bf16* zero,
                               const ConvShape& sh, long m, int kd) {
  if (m >= sh.M || kd >= sh.KD) return zero;
  if (MODE == MODE_FWD) {
    // m -> (n,p,q); kd = (r*S + s)*C + c
    const int q = (int)(m % sh.Q);
    long t = m / sh.Q;
    const int p = (int)(t % sh.P);
    const int n = (int)(t / sh.P);
    const int c = kd % sh.C;
    const int rs = kd / sh.C;
    const int s = rs % sh.S;
    const int r = rs / sh.S;
    const int h = p * sh.stride + r - sh.pad;
    const int w = q * sh.stride + s - sh.pad;
    if (h < 0 || h >= sh.H || w < 0 || w >= sh.W) return zero;
    return a + (((long)n * sh.H + h) * sh.W + w) * sh.C + c;
  } else {
    // m -> (n,h,w); kd = (r*S + s)*K + k; valid iff stride divides
    const int w = (int)(m % sh.W);
    long t = m / sh.W;
    const int h = (int)(t % sh.H);
    const int n = (int)(t / sh.H);
    const int k = kd % sh.K;
    const int rs = kd / sh.K;
    const int s = rs % sh.S;
    const int r = rs / sh.S;
    const int hp = h + sh.pad - r;
    const int wp = w + sh.pad - s;
    if (hp < 0 || wp < 0) return zero;
    if (hp % sh.stride || wp % sh.stride) return zero;
    const int p = hp / sh.stride, q = wp / sh.stride;
    if (p >= sh.P || q >= sh.Q) return zero;
    return a + (((long)n * sh.P + p) * sh.Q + q) * sh.K + k;
  }
}

// B matrix chunk: row j (output col), kd..kd+7 contiguous. B storage is
// (Nout, KD) row-major: W (K,RSC) for fwd, Wt (C,RSK) for bwd-data.
AL_DEV const bf16* b_chunk_ptr(const bf16* __restrict__ b, const bf16* zero,
                               const ConvShape& sh, int j, int kd) {
  if (j >= sh.Nout) return zero;
  return b + (long)j * sh.KD + kd;
}

// ---------------------------------------------------------------------------
// the tiled kernel
// ---------------------------------------------------------------------------

// GWR x GWC: wave grid (4 waves). (2,2) -> 128x128 tile; (4,1) -> 256x64 for
// narrow-Nout layers (K=64) where half a 128-wide tile would idle.
// epi_scale/epi_shift: optional fused inference epilogue
//   y = [relu]( acc * scale[col] + shift[col] [+ residual] )
// (frozen-stats BN folded into the conv — query/eval path)
template <int MODE, int GWR, int GWC>
__launch_bounds__(256)
__global__ void igemm_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                             bf16* __restrict__ out, const bf16* __restrict__ zero,
                             ConvShape sh, int grid_m,
                             const float* __restrict__ epi_scale,
                             const float* __restrict__ epi_shift,
                             const bf16* __restrict__ epi_res, int epi_relu,
                             float* __restrict__ stat_sum,
                             float* __restrict__ stat_sumsq,
                             const unsigned char* __restrict__ bnb_mask,
                             const bf16* __restrict__ bnb_x,
                             const float* __restrict__ bnb_mean,
                             const float* __restrict__ bnb_invstd) {
  constexpr int BM = GWR * 64, BN = GWC * 64, BK = 64;
  constexpr int NA = 2 * GWR, NB = 2 * GWC;  // 16B chunks per thread/operand
  // XCD-aware block remap (T1): contiguous output tiles on one XCD share B
  // panels in its L2. Bijective variant for any grid size.
  const int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    const int nx = 8;
    const int q = nwg / nx, r = nwg % nx;
    const int xcd = bid % nx, pos = bid / nx;
    if (pos < (xcd < r ? q + 1 : q))
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int bm = bid % grid_m, bn = bid / grid_m;
  const long m0 = (long)bm * BM;
  const int n0 = bn * BN;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* As = (bf16*)smem;                    // [2][BM][BK]
  bf16* Bs = As + 2 * BM * BK;               // [2][BN][BK]

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wr = wid / GWC, wc = wid % GWC;  // wave grid, 64x64 tiles each

  const int KT = sh.KD / BK;

  // -------------------------------------------------------------------
  // per-thread staging state, hoisted out of the K-loop: each thread owns
  // 4 A chunks and 4 B chunks per tile (16 B each). The GEMM row (and its
  // (n,p,q)/(n,h,w) decomposition) is FIXED per slot; only the contraction
  // coordinate advances (+BK per tile) with cheap carry propagation — no
  // divisions inside the loop.
  // -------------------------------------------------------------------
  int a_r[NA], a_s[NA], a_cf[NA];       // filter pos + fast-channel coord
  long a_pix[NA];                       // (n, outer-spatial) base index
  int a_p[NA], a_q[NA];                 // per-row spatial (fwd: p,q; bwd: h,w)
  bool a_ok[NA];
  const bf16* b_ptr[NB];
  int b_r[NB], b_s[NB], b_k[NB];        // MODE_BWD_S2: per-slot Wt coords
  long b_base[NB];
  const int fastC = (MODE == MODE_FWD) ? sh.C : sh.K;

#pragma unroll
  for (int i = 0; i < NA; ++i) {
    const int t = (wid * NA + i) * 64 + lane;
    const int row = t >> 3, u = t & 7;
    const int usw = u ^ (row & 7);
    const int kd0 = usw * 8;
    // contraction decomposition at kd0 (kd0 < 64 <= fastC except tails)
    int cf = kd0 % fastC;
    int rs = kd0 / fastC;
    a_cf[i] = cf;
    a_s[i] = rs % sh.S;
    a_r[i] = rs / sh.S;
    const long m = m0 + row;
    a_ok[i] = m < sh.M;
    if (a_ok[i]) {
      if (MODE == MODE_FWD) {
        const int q = (int)(m % sh.Q);
        long tt = m / sh.Q;
        const int p = (int)(tt % sh.P);
        const int n = (int)(tt / sh.P);
        a_pix[i] = (long)n * sh.H * sh.W;
        a_p[i] = p * sh.stride - sh.pad;
        a_q[i] = q * sh.stride - sh.pad;
      } else if (MODE == MODE_BWD_DATA) {
        const int w = (int)(m % sh.W);
        long tt = m / sh.W;
        const int h = (int)(tt % sh.H);
        const int n = (int)(tt / sh.H);
        a_pix[i] = (long)n * sh.P * sh.Q;
        a_p[i] = h + sh.pad;
        a_q[i] = w + sh.pad;
      } else {  // MODE_BWD_S2: m -> (n, hh, ww) within the parity class
        const int ww = (int)(m % sh.W);
        long tt = m / sh.W;
        const int hh = (int)(tt % sh.H);
        const int n = (int)(tt / sh.H);
        a_pix[i] = (long)n * sh.P * sh.Q;
        a_p[i] = hh + sh.poff;
        a_q[i] = ww + sh.qoff;
      }
    } else {
      a_pix[i] = 0; a_p[i] = 0; a_q[i] = 0;
    }
  }
#pragma unroll
  for (int i = 0; i < NB; ++i) {
    const int t = (wid * NB + i) * 64 + lane;
    const int row = t >> 3, u = t & 7;
    const int usw = u ^ (row & 7);
    // B: row n0 + row fixed; contraction contiguous except MODE_BWD_S2,
    // where only the class's (r,s) of Wt participate (strided, stateful)
    const int j = n0 + row;
    if (MODE == MODE_BWD_S2) {
      b_base[i] = (j < sh.Nout) ? (long)j * sh.b_rowstride : -1;
      b_r[i] = 0; b_s[i] = 0; b_k[i] = usw * 8;  // kd0 < 64 <= K
      b_ptr[i] = nullptr;
    } else {
      b_ptr[i] = (j < sh.Nout) ? (B + (long)j * sh.KD + usw * 8) : nullptr;
    }
  }

  auto a_src = [&](int i) -> const bf16* {
    if (!a_ok[i]) return zero;
    if (MODE == MODE_FWD) {
      const int h = a_p[i] + a_r[i];
      const int w = a_q[i] + a_s[i];
      if ((unsigned)h >= (unsigned)sh.H || (unsigned)w >= (unsigned)sh.W) return zero;
      return A + (a_pix[i] + (long)h * sh.W + w) * sh.C + a_cf[i];
    } else if (MODE == MODE_BWD_DATA) {
      const int hp = a_p[i] - a_r[i];
      const int wp = a_q[i] - a_s[i];
      if (hp < 0 || wp < 0) return zero;
      if (hp % sh.stride || wp % sh.stride) return zero;
      const int p = hp / sh.stride, q = wp / sh.stride;
      if (p >= sh.P || q >= sh.Q) return zero;
      return A + (a_pix[i] + (long)p * sh.Q + q) * sh.K + a_cf[i];
    } else {  // MODE_BWD_S2: every (ri, si) of the class divides exactly
      const int p = a_p[i] - a_r[i];
      const int q = a_q[i] - a_s[i];
      if ((unsigned)p >= (unsigned)sh.P || (unsigned)q >= (unsigned)sh.Q) return zero;
      return A + (a_pix[i] + (long)p * sh.Q + q) * sh.K + a_cf[i];
    }
  };

  auto b_src = [&](int i) -> const bf16* {
    if (MODE != MODE_BWD_S2) return b_ptr[i] ? b_ptr[i] : zero;
    if (b_base[i] < 0) return zero;
    return B + b_base[i] + sh.b_coff + (long)b_r[i] * sh.b_rstride
           + (long)b_s[i] * 2 * sh.K + b_k[i];
  };

  auto advance = [&]() {
#pragma unroll
    for (int i = 0; i < NA; ++i) {
      a_cf[i] += BK;
      while (a_cf[i] >= fastC) {
        a_cf[i] -= fastC;
        if (++a_s[i] == sh.S) { a_s[i] = 0; ++a_r[i]; }
      }
    }
#pragma unroll
    for (int i = 0; i < NB; ++i) {
      if (MODE == MODE_BWD_S2) {
        b_k[i] += BK;
        while (b_k[i] >= sh.K) {
          b_k[i] -= sh.K;
          if (++b_s[i] == sh.S) { b_s[i] = 0; ++b_r[i]; }
        }
      } else if (b_ptr[i]) {
        b_ptr[i] += BK;
      }
    }
  };

  auto stage = [&](int buf) {
    bf16* abase = As + buf * BM * BK;
    bf16* bbase = Bs + buf * BN * BK;
#pragma unroll
    for (int i = 0; i < NA; ++i) {
      const bf16* src = a_src(i);
      __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)src,
                                       (__attribute__((address_space(3))) void*)(abase + (wid * NA + i) * 512),
                                       16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < NB; ++i) {
      const bf16* src = b_src(i);
      __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)src,
                                       (__attribute__((address_space(3))) void*)(bbase + (wid * NB + i) * 512),
                                       16, 0, 0);
    }
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int l15 = lane & 15, l4 = lane >> 4;

  stage(0);
  int buf = 0;
  for (int kt = 0; kt < KT; ++kt) {
    // own-queue drain of tile kt, then one barrier; the NEXT tile's glds
    // (issued below) stay in flight across the whole compute phase — no
    // bottom barrier, so hipcc never emits a mid-loop vmcnt(0) drain.
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (kt + 1 < KT) {
      advance();
      stage(buf ^ 1);
    }
    const bf16* abase = As + buf * BM * BK;
    const bf16* bbase = Bs + buf * BN * BK;
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {  // two 32-deep chunks per K-tile
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
        const int arow = wr * 64 + fi * 16 + l15;
        const int aslot = (kc * 4 + l4) ^ (arow & 7);
        afrag[fi] = *(const bf16x8*)(abase + arow * BK + aslot * 8);
        const int brow = wc * 64 + fi * 16 + l15;
        const int bslot = (kc * 4 + l4) ^ (brow & 7);
        bfrag[fi] = *(const bf16x8*)(bbase + brow * BK + bslot * 8);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
    buf ^= 1;  // next iteration's top barrier orders buffer reuse
  }

  // ------------------------------------------------------------------
  // epilogue: the MFMA C/D fragment layout (row = (l>>4)*4 + r, col = l&15)
  // would store 16 scattered 2-byte elements per lane. Instead stage the
  // accumulator tile through LDS (reusing the A staging buffers) and emit
  // row-contiguous 16-byte stores — also making the fused-BN epilogue's
  // residual reads and scale/shift loads coalesced.
  // ------------------------------------------------------------------
  __syncthreads();  // all waves done reading As/Bs
  bf16* etile = As;  // [BM][BN] bf16 == 2*BM*BK elements (BN <= 2*BK)
  // optional per-column statistics for the following BatchNorm (training):
  // block-local sums in the free B-staging region, one global atomic per
  // column per block at the end (SURVEY.md §2.4 BN-stats fusion)
  // layout: [4 waves][CPR groups][8 cols] fp32, in the free B region
  float* sred = (float*)Bs;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        etile[(wr * 64 + mi * 16 + l4 * 4 + r) * BN + wc * 64 + ni * 16 + l15] =
            f2bf(acc[mi][ni][r]);
  __syncthreads();

  constexpr int CPR = BN / 8;              // 16B chunks per row
  constexpr int NCH = BM * CPR / 256;      // chunks per thread
  static_assert(256 % CPR == 0, "chunk->column-group invariant");
  float st_s[8] = {0, 0, 0, 0, 0, 0, 0, 0}, st_q[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
  for (int i = 0; i < NCH; ++i) {
    const int chunk = tid + 256 * i;
    const int row = chunk / CPR, cc = chunk % CPR;
    const long m = m0 + row;
    if (m >= sh.M) continue;
    const int col0 = n0 + cc * 8;
    if (col0 >= sh.Nout) continue;
    long off;
    if (MODE == MODE_BWD_S2) {
      const int ww = (int)(m % sh.W);
      long tt = m / sh.W;
      const int hh = (int)(tt % sh.H);
      const long n = tt / sh.H;
      off = ((n * sh.Horig + 2 * hh + sh.hcl) * sh.Worig + 2 * ww + sh.wcl)
                * sh.Nout + col0;
    } else {
      off = m * sh.Nout + col0;
    }
    s16x8 v = *(const s16x8*)(etile + row * BN + cc * 8);
    if (stat_sum) {
      // every chunk of this thread shares the same column group
      // (cc = tid % CPR since 256 % CPR == 0): accumulate in registers
      if (bnb_mask) {
        // BN-backward pre-reduce: this dx IS the upstream BatchNorm's dy.
        // Mask it in place and accumulate (sum dy~, sum dy~*xhat) so the
        // separate bn_bwd_reduce pass never runs (ops/functional.py).
        const unsigned char mb = bnb_mask[off >> 3];
        const s16x8 xv = *(const s16x8*)(bnb_x + off);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = ((mb >> j) & 1) && (col0 + j < sh.Nout) ? bits2f(v[j]) : 0.f;
          v[j] = f2bits(g);
          st_s[j] += g;
          if (col0 + j < sh.Nout)
            st_q[j] += g * (bits2f(xv[j]) - bnb_mean[col0 + j]) *
                       bnb_invstd[col0 + j];
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = (col0 + j < sh.Nout) ? bits2f(v[j]) : 0.f;
          st_s[j] += f;
          st_q[j] += f * f;
        }
      }
    }
    if (col0 + 8 <= sh.Nout) {
      if (epi_scale) {
        s16x8 rv;
        if (epi_res) rv = *(const s16x8*)(epi_res + off);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bits2f(v[j]) * epi_scale[col0 + j] + epi_shift[col0 + j];
          if (epi_res) f += bits2f(rv[j]);
          if (epi_relu) f = fmaxf(f, 0.f);
          v[j] = f2bits(f);
        }
      } else if (epi_res) {
        // plain residual-gradient accumulation (bwd-data): dx += dres,
        // replacing autograd's separate fan-in add pass
        const s16x8 rv = *(const s16x8*)(epi_res + off);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = f2bits(bits2f(v[j]) + bits2f(rv[j]));
      }
      *(s16x8*)(out + off) = v;
    } else {  // column tail: elementwise, all epi reads guarded
      for (int j = 0; col0 + j < sh.Nout; ++j) {
        float f = bits2f(v[j]);
        if (epi_scale) {
          f = f * epi_scale[col0 + j] + epi_shift[col0 + j];
          if (epi_res) f += bf2f(epi_res[off + j]);
          if (epi_relu) f = fmaxf(f, 0.f);
        } else if (epi_res) {
          f += bf2f(epi_res[off + j]);
        }
        short sj = f2bits(f);
        out[off + j] = *(bf16*)&sj;
      }
    }
  }
  if (stat_sum) {
    // lanes at stride CPR within a wave share a column group: shuffle-fold
    // them, then one LDS slot per (wave, group) and one global atomic per
    // column per block
#pragma unroll
    for (int j = 0; j < 8; ++j)
      for (int off = CPR; off < 64; off <<= 1) {
        st_s[j] += __shfl_xor(st_s[j], off, 64);
        st_q[j] += __shfl_xor(st_q[j], off, 64);
      }
    const int cc0 = tid % CPR;
    if (lane < CPR) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        sred[(wid * CPR + cc0) * 16 + j] = st_s[j];
        sred[(wid * CPR + cc0) * 16 + 8 + j] = st_q[j];
      }
    }
    __syncthreads();
    if (tid < BN && n0 + tid < sh.Nout) {
      const int cc = tid / 8, j = tid % 8;
      float ts = 0.f, tq = 0.f;
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        ts += sred[(w * CPR + cc) * 16 + j];
        tq += sred[(w * CPR + cc) * 16 + 8 + j];
      }
      atomicAdd(&stat_sum[n0 + tid], ts);
      atomicAdd(&stat_sumsq[n0 + tid], tq);
    }
  }
}

// ---------------------------------------------------------------------------
// 256x256 deep-pipelined conv GEMM: D[M, Nout] = A[M, KD] @ B[Nout, KD]^T.
// The guide's 8-phase 256^2 schedule: 8 waves (512 threads), BK=64, 128 KiB
// LDS as a ring of 4 half-tiles (128 rows) per operand, one half-tile
// refilled per phase into the slot freed by the previous phase, vmcnt(4)
// once per K-tile, no vmcnt(0) drain in the loop. Phase = one C-quadrant
// (qm,qn) x K=64: 12 ds_read_b128 + 2 glds + 16 MFMA.
//
// MODE_PURE: A is a plain (M, KD) row-major matrix — 1x1 s1 p0 convs (fwd:
// x with KD=C; bwd-data: dY with KD=K) need no im2col gather at all.
// MODE_FWD / MODE_BWD_DATA: A rows are im2col/scatter gathers exactly as in
// igemm_kernel, with the per-slot carry state (fixed GEMM row per slot, only
// the contraction coordinate advances — no divisions in the K-loop).
// ---------------------------------------------------------------------------

#define MODE_PURE 3

// BNT: output-tile width (256, or 128 for Nout%256!=0 layers e.g. the
// 128-wide layer2 convs). BMT: output-tile height; 512x128 restores the
// 256²-tile's MFMA density (128 block-MFMAs per phase) for gathered
// 128-wide layers, where 256x128's 64 could not cover the phase overhead.
template <int MODE, int BNT = 256, int BMT = 256>
__launch_bounds__(512, 1)
__global__ void gemm256_nt_kernel(const bf16* __restrict__ A,
                                  const bf16* __restrict__ B,
                                  bf16* __restrict__ out,
                                  const bf16* __restrict__ zero,
                                  ConvShape sh, int grid_m,
                                  const float* __restrict__ epi_scale,
                                  const float* __restrict__ epi_shift,
                                  const bf16* __restrict__ epi_res, int epi_relu,
                                  float* __restrict__ stat_sum,
                                  float* __restrict__ stat_sumsq,
                                  const unsigned char* __restrict__ bnb_mask,
                                  const bf16* __restrict__ bnb_x,
                                  const float* __restrict__ bnb_mean,
                                  const float* __restrict__ bnb_invstd) {
  const long M = sh.M;
  const int Nout = sh.Nout, KD = sh.KD;
  constexpr int BM = BMT, BN = BNT, BK = 64;
  constexpr int HALF = (BMT / 2) * BK;    // A half-tile elements
  constexpr int HALF_B = (BNT / 2) * BK;  // B half-tile elements
  constexpr int NLA = BMT / 128;          // A glds per thread per half
  constexpr int NLB = (BNT == 256) ? 2 : 1;  // B glds per thread per half
  constexpr int NF = BNT / 128;           // per-wave N fragments
  constexpr int MF = BMT / 64;            // per-wave M fragments
  constexpr int NSL = 2 * NLA;            // A gather slots (2 halves)
  // XCD-aware bijective block remap (T1)
  const int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    const int nx = 8;
    const int q = nwg / nx, r = nwg % nx;
    const int xcd = bid % nx, pos = bid / nx;
    if (pos < (xcd < r ? q + 1 : q))
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int bm = bid % grid_m, bn = bid / grid_m;
  const long m0 = (long)bm * BMT;
  const int n0 = bn * BN;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // slot layout: A half-tiles [buf][half] then B half-tiles [buf][half]
  bf16* As = (bf16*)smem;               // 4 x HALF
  bf16* Bs = As + 4 * HALF;             // 4 x HALF_B

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wrq = wid >> 2, wcq = wid & 3;  // wave grid inside a 128x128 quadrant
  const int l15 = lane & 15, l4 = lane >> 4;
  const int KT = KD / BK;

  // ---- per-slot A gather state (gather modes): each thread stages 2 chunks
  // per A half-tile refill; the GEMM row m of slot (h, li) is FIXED, so its
  // (n, spatial) decomposition is computed once and only the contraction
  // coordinate (r, s, cf) advances by +BK per refill — same carry scheme as
  // igemm_kernel, four independent slot states.
  const int fastC = (MODE == MODE_FWD) ? sh.C : sh.K;
  int ga_r[NSL], ga_s[NSL], ga_cf[NSL];
  long ga_pix[NSL];
  int ga_p[NSL], ga_q[NSL];
  bool ga_ok[NSL];
  if (MODE != MODE_PURE) {
#pragma unroll
    for (int sl = 0; sl < NSL; ++sl) {
      const int h = sl / NLA, li = sl % NLA;
      const int dc = li * 512 + wid * 64 + lane;
      const int drow = dc >> 3, du = dc & 7;
      const int su = du ^ (drow & 7);
      const int kd0 = su * 8;           // kd0 < 64 <= fastC
      ga_cf[sl] = kd0 % fastC;
      const int rs = kd0 / fastC;
      ga_s[sl] = rs % sh.S;
      ga_r[sl] = rs / sh.S;
      const long m = m0 + h * (BMT / 2) + drow;
      ga_ok[sl] = m < M;
      if (ga_ok[sl]) {
        if (MODE == MODE_FWD) {
          const int q = (int)(m % sh.Q);
          long tt = m / sh.Q;
          const int p = (int)(tt % sh.P);
          const int n = (int)(tt / sh.P);
          ga_pix[sl] = (long)n * sh.H * sh.W;
          ga_p[sl] = p * sh.stride - sh.pad;
          ga_q[sl] = q * sh.stride - sh.pad;
        } else {
          const int w = (int)(m % sh.W);
          long tt = m / sh.W;
          const int h_ = (int)(tt % sh.H);
          const int n = (int)(tt / sh.H);
          ga_pix[sl] = (long)n * sh.P * sh.Q;
          ga_p[sl] = h_ + sh.pad;
          ga_q[sl] = w + sh.pad;
        }
      } else {
        ga_pix[sl] = 0; ga_p[sl] = 0; ga_q[sl] = 0;
      }
    }
  }
  auto ga_src = [&](int sl) -> const bf16* {
    if (!ga_ok[sl]) return zero;
    if (MODE == MODE_FWD) {
      const int h = ga_p[sl] + ga_r[sl];
      const int w = ga_q[sl] + ga_s[sl];
      if ((unsigned)h >= (unsigned)sh.H || (unsigned)w >= (unsigned)sh.W) return zero;
      return A + (ga_pix[sl] + (long)h * sh.W + w) * sh.C + ga_cf[sl];
    } else {
      const int hp = ga_p[sl] - ga_r[sl];
      const int wp = ga_q[sl] - ga_s[sl];
      if (hp < 0 || wp < 0) return zero;
      if (hp % sh.stride || wp % sh.stride) return zero;
      const int p = hp / sh.stride, q = wp / sh.stride;
      if (p >= sh.P || q >= sh.Q) return zero;
      return A + (ga_pix[sl] + (long)p * sh.Q + q) * sh.K + ga_cf[sl];
    }
  };
  auto ga_advance = [&](int h) {      // +BK for the slots of half h
#pragma unroll
    for (int li = 0; li < NLA; ++li) {
      const int sl = h * NLA + li;
      ga_cf[sl] += BK;
      while (ga_cf[sl] >= fastC) {
        ga_cf[sl] -= fastC;
        if (++ga_s[sl] == sh.S) { ga_s[sl] = 0; ++ga_r[sl]; }
      }
    }
  };

  // ---- half-tile staging: chunk dc = load*512 + wid*64 + lane lands at LDS
  // byte dc*16 of the slot; the bank-conflict swizzle (slot ^= row&7, as in
  // igemm_kernel) is applied by remapping which SOURCE chunk each lane loads.
  // A-halves of gather modes read through the slot state (advance first!).
  auto stage_a = [&](int h, int ktile, bf16* slot) {
#pragma unroll
    for (int li = 0; li < NLA; ++li) {
      const bf16* src;
      if (MODE == MODE_PURE) {
        const int dc = li * 512 + wid * 64 + lane;
        const int drow = dc >> 3, du = dc & 7;
        const int su = du ^ (drow & 7);
        const long row = m0 + h * (BMT / 2) + drow;
        src = (ktile < KT && row < M) ? A + row * KD + (long)ktile * BK + su * 8
                                      : zero;
      } else {
        src = (ktile < KT) ? ga_src(h * NLA + li) : zero;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(slot + li * 512 * 8 + wid * 512),
          16, 0, 0);
    }
  };
  auto stage_b = [&](int h, int ktile, bf16* slot) {
#pragma unroll
    for (int li = 0; li < NLB; ++li) {
      const int dc = li * 512 + wid * 64 + lane;
      const int drow = dc >> 3, du = dc & 7;
      const int su = du ^ (drow & 7);
      const long row = n0 + h * (BNT / 2) + drow;
      const bf16* src = (ktile < KT && row < Nout)
                            ? B + row * KD + (long)ktile * BK + su * 8
                            : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(slot + li * 512 * 8 + wid * 512),
          16, 0, 0);
    }
  };
  // refill schedule within a K-tile group g (consuming buf = g&1): the q-order
  // (0,0),(0,1),(1,0),(1,1) frees one slot per phase; each phase refills the
  // slot last read in the previous phase:
  //   ph0: A[buf^1] h1 <- ktile g+1   ph1: B[buf^1] h1 <- g+1
  //   ph2: A[buf]   h0 <- ktile g+2   ph3: B[buf]   h0 <- g+2
  auto refill = [&](int g, int ph) {
    const int buf = g & 1;
    switch (ph) {
      case 0:
        if (MODE != MODE_PURE) ga_advance(1);
        stage_a(1, g + 1, As + ((buf ^ 1) * 2 + 1) * HALF);
        break;
      case 1: stage_b(1, g + 1, Bs + ((buf ^ 1) * 2 + 1) * HALF_B); break;
      case 2:
        if (MODE != MODE_PURE) ga_advance(0);
        stage_a(0, g + 2, As + (buf * 2) * HALF);
        break;
      case 3: stage_b(0, g + 2, Bs + (buf * 2) * HALF_B); break;
    }
  };

  f32x4 acc[2][2][MF][NF];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int g = 0; g < MF; ++g)
#pragma unroll
        for (int n = 0; n < NF; ++n) acc[i][j][g][n] = f32x4{0.f, 0.f, 0.f, 0.f};

  // prologue: K-tile 0 fully + K-tile 1's h0 halves (6 half-tiles); K-tile
  // 1's h1 halves arrive via group 0's ph0/ph1 refills. Order matters for
  // the vmcnt(4) accounting below. Gather-state after prologue: h0 @ kt1,
  // h1 @ kt0 — each subsequent refill advances its half by one K-tile.
  stage_a(0, 0, As);
  stage_b(0, 0, Bs);
  stage_a(1, 0, As + HALF);
  stage_b(1, 0, Bs + HALF_B);
  if (MODE != MODE_PURE) ga_advance(0);
  stage_a(0, 1, As + 2 * HALF);
  stage_b(0, 1, Bs + 2 * HALF_B);

  for (int g = 0; g < KT; ++g) {
    const int buf = g & 1;
    // once per K-tile: the 2 half-tiles issued since this tile's last half
    // (one A refill = NLA loads + one B refill = NLB) may stay in flight —
    // never drain to vmcnt(0) (T3+T4)
    if (NLA + NLB == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (NLA + NLB == 5) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
    else asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int qm = ph >> 1, qn = ph & 1;
      const bf16* ah = As + (buf * 2 + qm) * HALF;
      const bf16* bh = Bs + (buf * 2 + qn) * HALF_B;
      if constexpr (MF <= 4) {
        bf16x8 af[MF][2], bf[NF][2];
#pragma unroll
        for (int fg = 0; fg < MF; ++fg) {
          const int rowh = wrq * (BMT / 4) + fg * 16 + l15;
#pragma unroll
          for (int kc = 0; kc < 2; ++kc)
            af[fg][kc] = *(const bf16x8*)(ah + rowh * BK + (((kc * 4 + l4) ^ (rowh & 7)) * 8));
        }
#pragma unroll
        for (int ng = 0; ng < NF; ++ng) {
          const int colh = wcq * (BNT / 8) + ng * 16 + l15;
#pragma unroll
          for (int kc = 0; kc < 2; ++kc)
            bf[ng][kc] = *(const bf16x8*)(bh + colh * BK + (((kc * 4 + l4) ^ (colh & 7)) * 8));
        }
        refill(g, ph);
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kc = 0; kc < 2; ++kc)
#pragma unroll
          for (int fg = 0; fg < MF; ++fg)
#pragma unroll
            for (int ng = 0; ng < NF; ++ng)
              acc[qm][qn][fg][ng] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[fg][kc], bf[ng][kc], acc[qm][qn][fg][ng], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        __builtin_amdgcn_s_barrier();
      } else {
        // tall tile (MF=8): read fragments per 32-deep chunk so only MF+NF
        // frags are live at once (af[MF][2] would spill at 512 rows)
        refill(g, ph);
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          bf16x8 af[MF], bf[NF];
#pragma unroll
          for (int fg = 0; fg < MF; ++fg) {
            const int rowh = wrq * (BMT / 4) + fg * 16 + l15;
            af[fg] = *(const bf16x8*)(ah + rowh * BK + (((kc * 4 + l4) ^ (rowh & 7)) * 8));
          }
#pragma unroll
          for (int ng = 0; ng < NF; ++ng) {
            const int colh = wcq * (BNT / 8) + ng * 16 + l15;
            bf[ng] = *(const bf16x8*)(bh + colh * BK + (((kc * 4 + l4) ^ (colh & 7)) * 8));
          }
#pragma unroll
          for (int fg = 0; fg < MF; ++fg)
#pragma unroll
            for (int ng = 0; ng < NF; ++ng)
              acc[qm][qn][fg][ng] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[fg], bf[ng], acc[qm][qn][fg][ng], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
        __builtin_amdgcn_s_barrier();
      }
    }
  }

  // ---- epilogue: LDS round trip for contiguous 16B stores (same scheme as
  // igemm_kernel; etile = [256][256] bf16 = the whole 128 KiB LDS)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // tail zero-page glds
  __syncthreads();
  bf16* etile = As;
#pragma unroll
  for (int qm = 0; qm < 2; ++qm)
#pragma unroll
    for (int qn = 0; qn < 2; ++qn)
#pragma unroll
      for (int fg = 0; fg < MF; ++fg)
#pragma unroll
        for (int ng = 0; ng < NF; ++ng)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            etile[(qm * (BMT / 2) + wrq * (BMT / 4) + fg * 16 + l4 * 4 + r) * BN +
                  qn * (BNT / 2) + wcq * (BNT / 8) + ng * 16 + l15] =
                f2bf(acc[qm][qn][fg][ng][r]);
  __syncthreads();

  constexpr int CPR = BN / 8;          // 32 16B chunks per row
  constexpr int NCH = BM * CPR / 512;  // 16 chunks per thread
  static_assert(512 % CPR == 0, "chunk->column-group invariant");
  float st_s[8] = {0, 0, 0, 0, 0, 0, 0, 0}, st_q[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
  for (int i = 0; i < NCH; ++i) {
    const int chunk = tid + 512 * i;
    const int row = chunk / CPR, cc = chunk % CPR;
    const long m = m0 + row;
    if (m >= M) continue;
    const int col0 = n0 + cc * 8;     // Nout % BNT == 0: no column tail
    const long off = m * Nout + col0;
    s16x8 v = *(const s16x8*)(etile + row * BN + cc * 8);
    if (stat_sum) {
      if (bnb_mask) {  // BN-backward pre-reduce (see igemm_kernel epilogue)
        const unsigned char mb = bnb_mask[off >> 3];
        const s16x8 xv = *(const s16x8*)(bnb_x + off);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = (mb >> j) & 1 ? bits2f(v[j]) : 0.f;
          v[j] = f2bits(g);
          st_s[j] += g;
          st_q[j] += g * (bits2f(xv[j]) - bnb_mean[col0 + j]) *
                     bnb_invstd[col0 + j];
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = bits2f(v[j]);
          st_s[j] += f;
          st_q[j] += f * f;
        }
      }
    }
    if (epi_scale) {
      s16x8 rv;
      if (epi_res) rv = *(const s16x8*)(epi_res + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bits2f(v[j]) * epi_scale[col0 + j] + epi_shift[col0 + j];
        if (epi_res) f += bits2f(rv[j]);
        if (epi_relu) f = fmaxf(f, 0.f);
        v[j] = f2bits(f);
      }
    } else if (epi_res) {  // plain residual-grad accumulation (bwd-data)
      const s16x8 rv = *(const s16x8*)(epi_res + off);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v[j] = f2bits(bits2f(v[j]) + bits2f(rv[j]));
    }
    *(s16x8*)(out + off) = v;
  }
  if (stat_sum) {
    // lanes at stride CPR share a column group: xor folds, then the
    // per-wave partials go through LDS (etile reads are done) and one global
    // atomic per column per block
#pragma unroll
    for (int j = 0; j < 8; ++j)
      for (int off = CPR; off < 64; off <<= 1) {
        st_s[j] += __shfl_xor(st_s[j], off, 64);
        st_q[j] += __shfl_xor(st_q[j], off, 64);
      }
    __syncthreads();
    float* sred = (float*)As;         // [8 waves][CPR groups][16]
    const int cc0 = tid % CPR;
    if (lane < CPR) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        sred[(wid * CPR + cc0) * 16 + j] = st_s[j];
        sred[(wid * CPR + cc0) * 16 + 8 + j] = st_q[j];
      }
    }
    __syncthreads();
    if (tid < BN) {
      const int cc = tid / 8, j = tid % 8;
      float ts = 0.f, tq = 0.f;
#pragma unroll
      for (int w = 0; w < 8; ++w) {
        ts += sred[(w * CPR + cc) * 16 + j];
        tq += sred[(w * CPR + cc) * 16 + 8 + j];
      }
      atomicAdd(&stat_sum[n0 + tid], ts);
      atomicAdd(&stat_sumsq[n0 + tid], tq);
    }
  }
}

// ---------------------------------------------------------------------------
// direct fallback (any shape; used for the C=3 stems): one output element per
// thread, fp32 accumulate. Slow path by design — stems are ~3% of FLOPs.
// ---------------------------------------------------------------------------

template <int MODE>
__global__ void conv_direct_kernel(const bf16* __restrict__ A,
                                   const bf16* __restrict__ B,
                                   bf16* __restrict__ out, ConvShape sh,
                                   const float* __restrict__ epi_scale,
                                   const float* __restrict__ epi_shift,
                                   const bf16* __restrict__ epi_res, int epi_relu,
                                   float* __restrict__ stat_sum,
                                   float* __restrict__ stat_sumsq,
                                   const unsigned char* __restrict__ bnb_mask,
                                   const bf16* __restrict__ bnb_x,
                                   const float* __restrict__ bnb_mean,
                                   const float* __restrict__ bnb_invstd) {
  const long total = sh.M * sh.Nout;
  for (long i = grid_stride_begin(); i < total; i += grid_stride_step()) {
    const long m = i / sh.Nout;
    const int j = (int)(i % sh.Nout);
    float acc = 0.f;
    for (int kd = 0; kd < sh.KD; ++kd) {
      const bf16* pa = a_chunk_ptr<MODE>(A, nullptr, sh, m, kd);
      if (pa == nullptr) continue;
      acc += bf2f(*pa) * bf2f(B[(long)j * sh.KD + kd]);
    }
    if (epi_scale) {
      acc = acc * epi_scale[j] + epi_shift[j];
      if (epi_res) acc += bf2f(epi_res[i]);
      if (epi_relu) acc = fmaxf(acc, 0.f);
    } else if (epi_res) {
      acc += bf2f(epi_res[i]);
    }
    if (stat_sum) {
      atomicAdd(&stat_sum[j], acc);
      atomicAdd(&stat_sumsq[j], acc * acc);
    }
    out[i] = f2bf(acc);
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

static inline bool igemm_ok(int mode, const ConvShape& sh) {
  // 16B A-chunks must not straddle an (r,s) boundary: the fast gather dim
  // (C for fwd, K for bwd) must be a multiple of 8; contraction tiles exact.
  const int fast = (mode == MODE_FWD) ? sh.C : sh.K;
  return (sh.KD % 64 == 0) && (fast % 8 == 0);
}

extern "C" void al_conv2d_mm(int mode, const void* A, const void* B, void* out,
                             const void* zero_page, int N, int H, int W, int C,
                             int K, int R, int S, int P, int Q, int stride, int pad,
                             const float* epi_scale, const float* epi_shift,
                             const void* epi_res, int epi_relu, float* stat_sum,
                             float* stat_sumsq, const void* bnb_mask,
                             const void* bnb_x, const float* bnb_mean,
                             const float* bnb_invstd, hipStream_t stream) {
  ConvShape sh;
  sh.N = N; sh.H = H; sh.W = W; sh.C = C; sh.K = K; sh.R = R; sh.S = S;
  sh.P = P; sh.Q = Q; sh.stride = stride; sh.pad = pad;
  if (mode == MODE_FWD) {
    sh.M = (long)N * P * Q;
    sh.Nout = K;
    sh.KD = R * S * C;
  } else {
    sh.M = (long)N * H * W;
    sh.Nout = C;
    sh.KD = R * S * K;
  }

  // stride-2 bwd-data: run the 4 parity classes as dense gather GEMMs
  // (each contracts only its valid (r,s): no zero-chunk redundancy)
  if (mode == MODE_BWD_DATA && stride == 2 && (K % 64) == 0 && R * S > 1) {
    bool classes_ok = true;
    for (int hc = 0; hc < 2 && classes_ok; ++hc)
      for (int wc = 0; wc < 2 && classes_ok; ++wc) {
        const int rpar = (hc + pad) & 1, spar = (wc + pad) & 1;
        if ((R - rpar + 1) / 2 <= 0 || (S - spar + 1) / 2 <= 0) classes_ok = false;
      }
    if (classes_ok) {
      for (int hc = 0; hc < 2; ++hc)
        for (int wc = 0; wc < 2; ++wc) {
          ConvShape c = sh;
          const int rpar = (hc + pad) & 1, spar = (wc + pad) & 1;
          const int Rc = (R - rpar + 1) / 2, Sc = (S - spar + 1) / 2;
          const int Hc = (H - hc + 1) / 2, Wc = (W - wc + 1) / 2;
          if (Hc <= 0 || Wc <= 0) continue;
          c.H = Hc; c.W = Wc; c.R = Rc; c.S = Sc;
          c.M = (long)N * Hc * Wc;
          c.KD = Rc * Sc * K;
          c.Horig = H; c.Worig = W;
          c.hcl = hc; c.wcl = wc;
          c.poff = (hc + pad - rpar) / 2;
          c.qoff = (wc + pad - spar) / 2;
          c.b_rowstride = (long)R * S * K;
          c.b_rstride = 2L * S * K;
          c.b_coff = ((long)rpar * S + spar) * K;
          const bool narrow = c.Nout <= 64;
          const int BM = narrow ? 256 : 128, BN = narrow ? 64 : 128;
          const int gm = (int)((c.M + BM - 1) / BM);
          const int gn = (c.Nout + BN - 1) / BN;
          const size_t lds = 2 * (size_t)(BM + BN) * 64 * sizeof(bf16);
          dim3 grid(gm * gn), block(256);
          if (narrow)
            hipLaunchKernelGGL((igemm_kernel<MODE_BWD_S2, 4, 1>), grid, block, lds,
                               stream, (const bf16*)A, (const bf16*)B, (bf16*)out,
                               (const bf16*)zero_page, c, gm, epi_scale, epi_shift,
                               (const bf16*)epi_res, epi_relu, stat_sum, stat_sumsq, (const unsigned char*)bnb_mask, (const bf16*)bnb_x, bnb_mean, bnb_invstd);
          else
            hipLaunchKernelGGL((igemm_kernel<MODE_BWD_S2, 2, 2>), grid, block, lds,
                               stream, (const bf16*)A, (const bf16*)B, (bf16*)out,
                               (const bf16*)zero_page, c, gm, epi_scale, epi_shift,
                               (const bf16*)epi_res, epi_relu, stat_sum, stat_sumsq, (const unsigned char*)bnb_mask, (const bf16*)bnb_x, bnb_mean, bnb_invstd);
        }
      return;
    }
  }

  // 8-phase 256^2 route. 1x1 s1 p0 convs are plain NT GEMMs (MODE_PURE);
  // larger-filter s1 convs use the gathered variant when the grid still
  // fills the 256 CUs (1 block/CU at 128 KiB LDS: require >= min_grid
  // workgroups or the deep pipeline loses to the 128^2 tile's occupancy).
  static int gemm256_on = -1;
  static int min_grid = 192;
  if (gemm256_on < 0) {
    const char* e = getenv("AL_DISABLE_GEMM256");
    gemm256_on = (e && e[0] == '1') ? 0 : 1;
    const char* g = getenv("AL_GEMM256_MIN_GRID");
    if (g) min_grid = atoi(g);
  }
  // pure route: 1x1 s1 p0 only (contiguous rows). gather route: any stride
  // for fwd (the per-slot state uses p*stride - pad directly — this picks up
  // the 1x1-s2 downsamples and the stride-2 3x3s); bwd-data keeps stride 1
  // (stride-2 bwd goes through the parity-class decomposition above).
  if (gemm256_on && (mode == MODE_FWD || mode == MODE_BWD_DATA) &&
      (mode == MODE_FWD || stride == 1) &&
      sh.Nout % 128 == 0 && sh.KD % 64 == 0 && sh.M >= 128) {
    const int bnt = (sh.Nout % 256 == 0) ? 256 : 128;
    const bool pure = (R == 1 && S == 1 && stride == 1 && pad == 0);
    const int fast = (mode == MODE_FWD) ? sh.C : sh.K;
    // tile height: the 512-row tall tile restores the 256² tile's per-phase
    // MFMA density for gathered 128-wide layers, but MEASURED ~20% slower
    // than the 2-barrier 128² kernel anyway (per-kc fragment re-reads double
    // the LDS traffic and the gather state grows to 8 slots) — kept for
    // round-2 tuning behind AL_GEMM256_TALL=1. Pure bwd-data at 128 wide
    // keeps the 256-row tile (measured +16%).
    static int tall = -1;
    if (tall < 0) {
      const char* e = getenv("AL_GEMM256_TALL");
      tall = (e && e[0] == '1') ? 1 : 0;
    }
    const int bmt = (tall && bnt == 128 && !pure) ? 512 : 256;
    const int grid_m = (int)((sh.M + bmt - 1) / bmt);
    const int grid_n = sh.Nout / bnt;
    const bool gather_ok = !pure && fast % 8 == 0 &&
                           grid_m * grid_n >= min_grid / (bmt / 256);
    const bool bnt_ok = (bnt == 256) ||
                        (pure && mode == MODE_BWD_DATA) ||
                        (tall && !pure);
    if ((pure || gather_ok) && bnt_ok) {
      static bool attr_set = false;
      if (!attr_set) {
#define SET_ATTR(MODE_, BNT_, BMT_) (void)hipFuncSetAttribute( \
        (const void*)gemm256_nt_kernel<MODE_, BNT_, BMT_>, \
        hipFuncAttributeMaxDynamicSharedMemorySize, 163840)
        SET_ATTR(MODE_PURE, 256, 256); SET_ATTR(MODE_FWD, 256, 256);
        SET_ATTR(MODE_BWD_DATA, 256, 256);
        SET_ATTR(MODE_PURE, 128, 256);
        SET_ATTR(MODE_FWD, 128, 512); SET_ATTR(MODE_BWD_DATA, 128, 512);
#undef SET_ATTR
        attr_set = true;
      }
      // LDS: A ring 4 * (bmt/2) * 64 * 2B + B ring 4 * (bnt/2) * 64 * 2B
      const size_t lds = 4 * (size_t)(bmt / 2) * 64 * 2 +
                         4 * (size_t)(bnt / 2) * 64 * 2;
      dim3 grid(grid_m * grid_n), block(512);
#define LAUNCH256(MODE_, BNT_, BMT_) hipLaunchKernelGGL( \
      (gemm256_nt_kernel<MODE_, BNT_, BMT_>), \
      grid, block, lds, stream, (const bf16*)A, (const bf16*)B, (bf16*)out, \
      (const bf16*)zero_page, sh, grid_m, epi_scale, epi_shift, \
      (const bf16*)epi_res, epi_relu, stat_sum, stat_sumsq, \
      (const unsigned char*)bnb_mask, (const bf16*)bnb_x, bnb_mean, bnb_invstd)
      if (bnt == 256) {
        if (pure) LAUNCH256(MODE_PURE, 256, 256);
        else if (mode == MODE_FWD) LAUNCH256(MODE_FWD, 256, 256);
        else LAUNCH256(MODE_BWD_DATA, 256, 256);
      } else if (pure) {
        LAUNCH256(MODE_PURE, 128, 256);
      } else if (mode == MODE_FWD) {
        LAUNCH256(MODE_FWD, 128, 512);
      } else {
        LAUNCH256(MODE_BWD_DATA, 128, 512);
      }
#undef LAUNCH256
      return;
    }
  }

  if (igemm_ok(mode, sh)) {
    // narrow-Nout layers (K=64) use a 256x64 tile so no wave idles
    const bool narrow = sh.Nout <= 64;
    const int BM = narrow ? 256 : 128, BN = narrow ? 64 : 128;
    const int grid_m = (int)((sh.M + BM - 1) / BM);
    const int grid_n = (sh.Nout + BN - 1) / BN;
    const size_t lds = 2 * (size_t)(BM + BN) * 64 * sizeof(bf16);
    dim3 grid(grid_m * grid_n), block(256);
#define LAUNCH(MODE_, GWR_, GWC_) hipLaunchKernelGGL((igemm_kernel<MODE_, GWR_, GWC_>), grid, block, lds, stream, (const bf16*)A, (const bf16*)B, (bf16*)out, (const bf16*)zero_page, sh, grid_m, epi_scale, epi_shift, (const bf16*)epi_res, epi_relu, stat_sum, stat_sumsq, \
      (const unsigned char*)bnb_mask, (const bf16*)bnb_x, bnb_mean, bnb_invstd)
    if (mode == MODE_FWD) {
      if (narrow) LAUNCH(MODE_FWD, 4, 1); else LAUNCH(MODE_FWD, 2, 2);
    } else {
      if (narrow) LAUNCH(MODE_BWD_DATA, 4, 1); else LAUNCH(MODE_BWD_DATA, 2, 2);
    }
#undef LAUNCH
  } else {
    long total = sh.M * sh.Nout;
    int blocks = (int)min((total + 255) / 256, (long)8192);
    if (mode == MODE_FWD)
      hipLaunchKernelGGL((conv_direct_kernel<MODE_FWD>), dim3(blocks), dim3(256), 0,
                         stream, (const bf16*)A, (const bf16*)B, (bf16*)out, sh,
                         epi_scale, epi_shift, (const bf16*)epi_res, epi_relu, stat_sum, stat_sumsq, (const unsigned char*)bnb_mask, (const bf16*)bnb_x, bnb_mean, bnb_invstd);
    else
      hipLaunchKernelGGL((conv_direct_kernel<MODE_BWD_DATA>), dim3(blocks), dim3(256),
                         0, stream, (const bf16*)A, (const bf16*)B, (bf16*)out, sh,
                         epi_scale, epi_shift, (const bf16*)epi_res, epi_relu, stat_sum, stat_sumsq, (const unsigned char*)bnb_mask, (const bf16*)bnb_x, bnb_mean, bnb_invstd);
  }
}

// ---------------------------------------------------------------------------
// strided scatter for the 1x1-stride-2 bwd-data route (ResNet downsample):
// dx[n, h, w, c] = (h % s == 0 && w % s == 0) ? tmp[n, h/s, w/s, c] : 0
// — one write pass replacing ATen's zeros-fill + strided-copy pair
// (functional.py previously: torch.zeros + dx[:, ::2, ::2] = tmp).
// ---------------------------------------------------------------------------

__global__ void scatter_s2_kernel(const bf16* __restrict__ tmp,
                                  bf16* __restrict__ dx, int N, int H, int W,
                                  int C, int P, int Q, int stride) {
  // one 16B chunk (8 bf16, C % 8 == 0) per iteration, fully coalesced
  const long chunks = (long)N * H * W * (C / 8);
  for (long i = grid_stride_begin(); i < chunks; i += grid_stride_step()) {
    const int cc = (int)(i % (C / 8));
    long t = i / (C / 8);
    const int w = (int)(t % W);
    t /= W;
    const int h = (int)(t % H);
    const long n = t / H;
    s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (h % stride == 0 && w % stride == 0) {
      const int p = h / stride, q = w / stride;
      if (p < P && q < Q)
        v = *(const s16x8*)(tmp + (((n * P + p) * Q + q) * C) + cc * 8);
    }
    *(s16x8*)(dx + i * 8) = v;
  }
}

extern "C" void al_scatter_s2(const void* tmp, void* dx, int N, int H, int W,
                              int C, int P, int Q, int stride,
                              hipStream_t stream) {
  const long chunks = (long)N * H * W * (C / 8);
  int blocks = (int)min((chunks + 255) / 256, (long)8192);
  hipLaunchKernelGGL(scatter_s2_kernel, dim3(blocks), dim3(256), 0, stream,
                     (const bf16*)tmp, (bf16*)dx, N, H, W, C, P, Q, stride);
}

// ---------------------------------------------------------------------------
// im2col packing for shapes the MFMA kernel cannot gather directly (the C=3
// stems): materialize A = im2col(x) zero-padded to KDpad columns (KDpad % 64
// == 0); the conv then runs as a 1x1 MFMA igemm over the packed buffer.
// Memory-bound; output writes are fully coalesced (consecutive threads ->
// consecutive kd).
// ---------------------------------------------------------------------------

__global__ void im2col_pack_kernel(const bf16* __restrict__ x, bf16* __restrict__ out,
                                   ConvShape sh, int kdpad, int rowpad, int c8n,
                                   int dc8, int dq, int dpr, int dpq,
                                   unsigned mr, unsigned mc) {
  // row-padded layout: kd = r*rowpad + (s*C + c), each filter row padded to
  // an 8-aligned rowpad so every 8-element chunk lies within ONE (r)-row.
  // One thread per 8-chunk, lanes CONTIGUOUS over the packed output (the
  // c8-in-blockIdx.y variant strided lane stores by c8n*16B and measured 3x
  // slower). v3 keeps v2's exact memory pattern but removes the three
  // per-item 64-bit divmods (~300 cycles/lane per 16B store — v2 measured
  // division-bound at 2.0 TB/s): (c8, q, p, n) advance by host-precomputed
  // carry chains (dc8 = step%c8n, dq = (step/c8n)%Q, dpr/dpq likewise) and
  // r/off0 come from exact 32-bit magic reciprocals (__umulhi; guaranteed
  // exact for divisor<=2^16 and dividend<=2^16).
  const int sc = sh.S * sh.C;
  const int kdrp = sh.R * rowpad;
  const long total = sh.M * (long)c8n;
  const long i0 = grid_stride_begin();
  if (i0 >= total) return;
  const long step = grid_stride_step();
  // one-time decode; afterwards carries only
  int c8 = (int)(i0 % c8n);
  long m = i0 / c8n;
  int q = (int)(m % sh.Q);
  long t = m / sh.Q;
  int pp = (int)(t % sh.P);
  int n = (int)(t / sh.P);
  for (long i = i0; i < total; i += step) {
    const int kd0 = c8 * 8;
    s16x8 o = {0, 0, 0, 0, 0, 0, 0, 0};
    if (kd0 < kdrp) {
      const int r = (int)__umulhi((unsigned)kd0, mr);
      const int off0 = kd0 - r * rowpad;
      if (off0 < sc) {
        const int h = pp * sh.stride - sh.pad + r;
        if ((unsigned)h < (unsigned)sh.H) {
          const int w0 = q * sh.stride - sh.pad;
          const bf16* src = x + (((long)n * sh.H + h) * sh.W + w0) * sh.C + off0;
          int wi = (int)__umulhi((unsigned)off0, mc) + w0;
          int c = off0 - (wi - w0) * sh.C;
          short* os = (short*)&o;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (off0 + j < sc && (unsigned)wi < (unsigned)sh.W)
              os[j] = *(const short*)&src[j];
            if (++c == sh.C) { c = 0; ++wi; }
          }
        }
      }
    }
    // nontemporal: apack is ~16x the input (1.2 GB at the B=256 stem), far
    // beyond L2 — don't let the stream evict the x rows being gathered
    __builtin_nontemporal_store(o, (s16x8*)out + i);
    // advance (c8, q, p, n) by `step` items without dividing
    c8 += dc8;
    int mc_ = c8 >= c8n;
    if (mc_) c8 -= c8n;
    q += dq + mc_;
    int cq = q >= sh.Q;
    if (cq) q -= sh.Q;
    pp += dpr + cq;
    int cp = pp >= sh.P;
    if (cp) pp -= sh.P;
    n += dpq + cp;
  }
}

extern "C" void al_im2col_pack(const void* x, void* out, int N, int H, int W, int C,
                               int R, int S, int P, int Q, int stride, int pad,
                               int kdpad, int rowpad, hipStream_t stream) {
  ConvShape sh;
  sh.N = N; sh.H = H; sh.W = W; sh.C = C; sh.K = 0; sh.R = R; sh.S = S;
  sh.P = P; sh.Q = Q; sh.stride = stride; sh.pad = pad;
  sh.M = (long)N * P * Q;
  sh.Nout = 0;
  sh.KD = R * S * C;
  const int c8n = kdpad / 8;
  long total = sh.M * (long)c8n;
  int blocks = (int)min((total + 255) / 256, (long)16384);
  const long step = (long)blocks * 256;
  // carry-chain increments for one grid-stride step (see kernel comment)
  const int dc8 = (int)(step % c8n);
  const long dm = step / c8n;
  const int dq = (int)(dm % Q);
  const long dp = dm / Q;
  const int dpr = (int)(dp % P), dpq = (int)(dp / P);
  // exact 32-bit reciprocals: ceil(2^32/d) is exact for d, n <= 2^16
  const unsigned mr = (unsigned)((0x100000000ULL + rowpad - 1) / rowpad);
  const unsigned mc = (unsigned)((0x100000000ULL + C - 1) / C);
  hipLaunchKernelGGL(im2col_pack_kernel, dim3(blocks), dim3(256), 0, stream,
                     (const bf16*)x, (bf16*)out, sh, kdpad, rowpad, c8n, dc8,
                     dq, dpr, dpq, mr, mc);
}
