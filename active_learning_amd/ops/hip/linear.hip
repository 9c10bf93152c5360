// First-party linear-head GEMM (fp32) on the exact-f32 MFMA.
//
// The classifier head (resnet_simclr.py:22,33,38) is (B,M)x(M,C) with
// B<=256, M<=2048, C<=1000 — tiny next to the convs, but it is the last
// library GEMM on the hot path (COVERAGE row 68). CDNA4 has an EXACT
// fp32-input MFMA (mfma_f32_16x16x4f32, guide §3: 157 TF rate, identical
// numerics to VALU fp32), so the head and both its backward GEMMs run
// first-party without any precision change vs the reference's fp32 Linear.
//
// One templated kernel serves the three contractions:
//   fwd: out(B,C) = X(B,M) @ W(C,M)^T  -> TA=0 TB=1 (+bias)
//   dx:  out(B,M) = dY(B,C) @ W(C,M)   -> TA=0 TB=0
//   dw:  out(C,M) = dY^T(C,B) @ X(B,M) -> TA=1 TB=0
// 64x64 tiles, 4 waves (2x2, 32x32 each), 64-deep fp32 LDS staging with a
// +1-float row pad (stride 65 => conflict-free strided access).

#include "al_common.h"

constexpr int LDP = 65;  // padded LDS row stride (floats)

// ZSPLIT: blockIdx.y strides the contraction so small (B,C) outputs still
// fill the 256 CUs; partial tiles accumulate with fp32 atomics into a
// zeroed output (bias applied by the first K-slice only).
template <bool TA, bool TB, bool BIAS, bool ZSPLIT = false>
__launch_bounds__(256)
__global__ void linear_gemm_kernel(const float* __restrict__ A,
                                   const float* __restrict__ Bm,
                                   const float* __restrict__ bias,
                                   float* __restrict__ out,
                                   int M_, int N_, int K_, int lda, int ldb,
                                   int grid_m) {
  __shared__ float As[64 * LDP];
  __shared__ float Bs[64 * LDP];
  const int bm = blockIdx.x % grid_m, bn = blockIdx.x / grid_m;
  const int r0 = bm * 64, n0 = bn * 64;
  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wr = wid >> 1, wc = wid & 1;
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kz0 = ZSPLIT ? (int)blockIdx.y * 64 : 0;
  const int kstep = ZSPLIT ? (int)gridDim.y * 64 : 64;
  for (int k0 = kz0; k0 < K_; k0 += kstep) {
    // stage A tile [row][k]
    for (int idx = tid; idx < 64 * 64; idx += 256) {
      int r, k;
      if (TA) { k = idx >> 6; r = idx & 63; }          // k-major global reads
      else    { r = idx >> 6; k = idx & 63; }
      const int gr = r0 + r, gk = k0 + k;
      float v = 0.f;
      if (gr < M_ && gk < K_)
        v = TA ? A[(long)gk * lda + gr] : A[(long)gr * lda + gk];
      As[r * LDP + k] = v;
    }
    // stage B tile [k][n]
    for (int idx = tid; idx < 64 * 64; idx += 256) {
      int k, n;
      if (TB) { n = idx >> 6; k = idx & 63; }          // n-major global reads
      else    { k = idx >> 6; n = idx & 63; }
      const int gn = n0 + n, gk = k0 + k;
      float v = 0.f;
      if (gn < N_ && gk < K_)
        v = TB ? Bm[(long)gn * ldb + gk] : Bm[(long)gk * ldb + gn];
      Bs[k * LDP + n] = v;
    }
    __syncthreads();
#pragma unroll 4
    for (int kk = 0; kk < 16; ++kk) {
      float a[2], b[2];
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        a[fi] = As[(wr * 32 + fi * 16 + l15) * LDP + kk * 4 + l4];
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        b[fj] = Bs[(kk * 4 + l4) * LDP + wc * 32 + fj * 16 + l15];
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a[fi], b[fj], acc[fi][fj], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int fi = 0; fi < 2; ++fi)
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
      const int col = n0 + wc * 32 + fj * 16 + l15;
      if (col >= N_) continue;
      const float bv = (BIAS && (!ZSPLIT || blockIdx.y == 0)) ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = r0 + wr * 32 + fi * 16 + l4 * 4 + r;
        if (row >= M_) continue;
        if (ZSPLIT) atomicAdd(&out[(long)row * N_ + col], acc[fi][fj][r] + bv);
        else out[(long)row * N_ + col] = acc[fi][fj][r] + bv;
      }
    }
}

// ---------------------------------------------------------------------------
// Fused BADGE gradient-embedding pairwise distances:
//   out[i][j] = d[i] + d[j] - 2 * (a_i . a_j) * (e_i . e_j)
// (<g_i,g_j> = (a_i.a_j)(e_i.e_j) for g = a (x) e — the (B, C*M) embedding
// never exists; reference materializes it, badge_sampler.py:36-48.)
// The round-1 torch composition (two rocBLAS skinny GEMMs + 4 elementwise
// passes per row chunk) re-streamed the N x chunk fp32 intermediates ~8x;
// this kernel computes both rank-K grams on f32 MFMA from LDS tiles and
// writes the N x N output ONCE. 64x64 tile per block, 4 waves (2x2).
// ---------------------------------------------------------------------------

__launch_bounds__(256)
__global__ void badge_gram_kernel(const float* __restrict__ Av,
                                  const float* __restrict__ Ev,
                                  const float* __restrict__ d,
                                  float* __restrict__ out,
                                  long N, int Ka, int Ke) {
  __shared__ float As[64 * LDP];
  __shared__ float Bs[64 * LDP];
  // grid-stride over output tiles: at N=130k there are ~4.1M 64x64 tiles,
  // and one-workgroup-per-tile is DISPATCH-bound (~2 s of launch overhead
  // measured); ~16k persistent workgroups loop instead
  const long gm = (N + 63) / 64;
  for (long t = blockIdx.x; t < gm * gm; t += gridDim.x) {
  const long r0 = (t % gm) * 64, n0 = (t / gm) * 64;
  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wr = wid >> 1, wc = wid & 1;
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 accA[2][2], accE[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      accA[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
      accE[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
    }
  __syncthreads();  // previous tile's epilogue reads done before restaging

  // one gram accumulation: rows from src[r0..], cols from src[n0..]
  auto gram = [&](const float* src, int K_, f32x4 (*acc)[2]) {
    for (int k0 = 0; k0 < K_; k0 += 64) {
      // only the live K columns are staged and contracted (the pooled BADGE
      // factors are K=16/32: a fixed 64-deep loop would 4x the work)
      const int kpad = min(64, ((K_ - k0) + 3) & ~3);
      for (int idx = tid; idx < 64 * 64; idx += 256) {
        const int r = idx >> 6, k = idx & 63;
        if (k >= kpad) continue;
        const long gr = r0 + r, gn = n0 + r;
        const int gk = k0 + k;
        As[r * LDP + k] = (gr < N && gk < K_) ? src[gr * K_ + gk] : 0.f;
        Bs[r * LDP + k] = (gn < N && gk < K_) ? src[gn * K_ + gk] : 0.f;
      }
      __syncthreads();
#pragma unroll 4
      for (int kk = 0; kk < kpad / 4; ++kk) {
        float a[2], b[2];
#pragma unroll
        for (int fi = 0; fi < 2; ++fi)
          a[fi] = As[(wr * 32 + fi * 16 + l15) * LDP + kk * 4 + l4];
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          b[fj] = Bs[(wc * 32 + fj * 16 + l15) * LDP + kk * 4 + l4];
#pragma unroll
        for (int fi = 0; fi < 2; ++fi)
#pragma unroll
          for (int fj = 0; fj < 2; ++fj)
            acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                a[fi], b[fj], acc[fi][fj], 0, 0, 0);
      }
      __syncthreads();
    }
  };
  gram(Av, Ka, (f32x4(*)[2])accA);
  gram(Ev, Ke, (f32x4(*)[2])accE);

#pragma unroll
  for (int fi = 0; fi < 2; ++fi)
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
      const long col = n0 + wc * 32 + fj * 16 + l15;
      if (col >= N) continue;
      const float dj = d[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long row = r0 + wr * 32 + fi * 16 + l4 * 4 + r;
        if (row < N)
          out[row * N + col] =
              d[row] + dj - 2.f * accA[fi][fj][r] * accE[fi][fj][r];
      }
    }
  }  // tile loop
}

extern "C" void al_badge_gram(const float* a, const float* e, const float* d,
                              float* out, long N, int Ka, int Ke,
                              hipStream_t stream) {
  const long g = (N + 63) / 64;
  const long blocks = min(g * g, (long)16384);
  hipLaunchKernelGGL(badge_gram_kernel, dim3((unsigned)blocks), dim3(256), 0,
                     stream, a, e, d, out, N, Ka, Ke);
}

// ---------------------------------------------------------------------------
// Pairwise squared distances on bf16 MFMA:
//   out[i][j] = sq[i] + sq[j] - 2 * <f_i, f_j>     (out fp32, f bf16)
// The coreset N x N matrix (68 GB at N=130k, resident in HBM). The fp32
// rocBLAS composition runs at the fp32 MFMA rate (~115 TF); the dot products
// here run on mfma_f32_16x16x32_bf16 with fp32 accumulation and fp32
// norms/out (PARITY.md: the embeddings themselves are bf16-computed).
// 128x128x64 tile, 4 waves (2x2), 2-barrier double-buffered glds staging
// with the igemm XOR source swizzle; grid-strided over output tiles.
// ---------------------------------------------------------------------------

typedef __bf16 pw_bf16x8 __attribute__((ext_vector_type(8)));

__launch_bounds__(256)
__global__ void pairwise_kernel(const bf16* __restrict__ F,
                                const float* __restrict__ sq,
                                float* __restrict__ out,
                                const bf16* __restrict__ zero,
                                long N, int M) {
  extern __shared__ __attribute__((aligned(16))) char pw_smem[];
  bf16* As = (bf16*)pw_smem;         // [2][128][64]
  bf16* Bs = As + 2 * 128 * 64;
  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wr = wid >> 1, wc = wid & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  const int KT = M / 64;
  const long gm = (N + 127) / 128;

  for (long t = blockIdx.x; t < gm * gm; t += gridDim.x) {
    const long r0 = (t % gm) * 128, n0 = (t / gm) * 128;
    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
    __syncthreads();  // previous tile's reads complete before restaging

    auto stage = [&](int buf, int kt) {
      bf16* ab = As + buf * 128 * 64;
      bf16* bb = Bs + buf * 128 * 64;
#pragma unroll
      for (int i = 0; i < 4; ++i) {  // 4 A-chunks + 4 B-chunks per thread
        const int tt = (wid * 4 + i) * 64 + lane;
        const int row = tt >> 3, u = tt & 7;
        const int usw = u ^ (row & 7);
        const long ra = r0 + row, rb = n0 + row;
        const bf16* sa = (ra < N) ? F + ra * M + (long)kt * 64 + usw * 8 : zero;
        const bf16* sb = (rb < N) ? F + rb * M + (long)kt * 64 + usw * 8 : zero;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)sa,
            (__attribute__((address_space(3))) void*)(ab + (wid * 4 + i) * 512),
            16, 0, 0);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)sb,
            (__attribute__((address_space(3))) void*)(bb + (wid * 4 + i) * 512),
            16, 0, 0);
      }
    };

    stage(0, 0);
    int buf = 0;
    for (int kt = 0; kt < KT; ++kt) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
      if (kt + 1 < KT) stage(buf ^ 1, kt + 1);
      const bf16* ab = As + buf * 128 * 64;
      const bf16* bb = Bs + buf * 128 * 64;
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        pw_bf16x8 af[4], bf[4];
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          const int ar = wr * 64 + f * 16 + l15;
          af[f] = *(const pw_bf16x8*)(ab + ar * 64 + (((kc * 4 + l4) ^ (ar & 7)) * 8));
          const int br = wc * 64 + f * 16 + l15;
          bf[f] = *(const pw_bf16x8*)(bb + br * 64 + (((kc * 4 + l4) ^ (br & 7)) * 8));
        }
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      }
      buf ^= 1;
    }

#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const long col = n0 + wc * 64 + ni * 16 + l15;
        if (col >= N) continue;
        const float sc = sq[col];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const long row = r0 + wr * 64 + mi * 16 + l4 * 4 + r;
          if (row < N)
            out[row * N + col] = sq[row] + sc - 2.f * acc[mi][ni][r];
        }
      }
  }
}

extern "C" int al_pairwise_sqdist(const void* f, const float* sq, float* out,
                                  const void* zero, long N, int M,
                                  hipStream_t stream) {
  if (M % 64 != 0) return -1;
  const long gm = (N + 127) / 128;
  const long blocks = min(gm * gm, (long)8192);
  const size_t lds = 4 * 128 * 64 * sizeof(bf16);  // 64 KiB (2 bufs x A+B)
  hipLaunchKernelGGL(pairwise_kernel, dim3((unsigned)blocks), dim3(256), lds,
                     stream, (const bf16*)f, sq, out, (const bf16*)zero, N, M);
  return 0;
}

// column sums of dY (B,C) -> db (C)
__global__ void colsum_kernel(const float* __restrict__ dy, float* __restrict__ db,
                              int B, int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f;
  for (int b = 0; b < B; ++b) s += dy[(long)b * C + c];
  db[c] = s;
}

static inline int lin_zsplit(int tiles, int K_) {
  int z = 256 / (tiles < 1 ? 1 : tiles);
  const int kt = (K_ + 63) / 64;
  if (z > kt) z = kt;
  return z < 1 ? 1 : z;
}

// whether the launcher will K-split (caller must then pass a ZEROED out)
extern "C" int al_linear_needs_zero(int M_, int N_, int K_) {
  return lin_zsplit(((M_ + 63) / 64) * ((N_ + 63) / 64), K_) > 1;
}

extern "C" void al_linear_fwd(const float* x, const float* w, const float* bias,
                              float* out, int B, int M, int C,
                              hipStream_t stream) {
  const int gm = (B + 63) / 64, gn = (C + 63) / 64;
  const int z = lin_zsplit(gm * gn, M);
  dim3 grid(gm * gn, z);
  if (z > 1) {
    if (bias)
      hipLaunchKernelGGL((linear_gemm_kernel<false, true, true, true>), grid,
                         dim3(256), 0, stream, x, w, bias, out, B, C, M, M, M, gm);
    else
      hipLaunchKernelGGL((linear_gemm_kernel<false, true, false, true>), grid,
                         dim3(256), 0, stream, x, w, nullptr, out, B, C, M, M, M, gm);
  } else if (bias) {
    hipLaunchKernelGGL((linear_gemm_kernel<false, true, true>), grid,
                       dim3(256), 0, stream, x, w, bias, out, B, C, M, M, M, gm);
  } else {
    hipLaunchKernelGGL((linear_gemm_kernel<false, true, false>), grid,
                       dim3(256), 0, stream, x, w, nullptr, out, B, C, M, M, M, gm);
  }
}

extern "C" void al_linear_bwd(const float* dy, const float* x, const float* w,
                              float* dx, float* dw, float* db, int B, int M,
                              int C, hipStream_t stream) {
  if (dx) {
    const int gm = (B + 63) / 64, gn = (M + 63) / 64;
    const int z = lin_zsplit(gm * gn, C);
    dim3 grid(gm * gn, z);
    if (z > 1)
      hipLaunchKernelGGL((linear_gemm_kernel<false, false, false, true>), grid,
                         dim3(256), 0, stream, dy, w, nullptr, dx, B, M, C, C, M, gm);
    else
      hipLaunchKernelGGL((linear_gemm_kernel<false, false, false>), grid,
                         dim3(256), 0, stream, dy, w, nullptr, dx, B, M, C, C, M, gm);
  }
  if (dw) {
    const int gm = (C + 63) / 64, gn = (M + 63) / 64;
    const int z = lin_zsplit(gm * gn, B);
    dim3 grid(gm * gn, z);
    if (z > 1)
      hipLaunchKernelGGL((linear_gemm_kernel<true, false, false, true>), grid,
                         dim3(256), 0, stream, dy, x, nullptr, dw, C, M, B, C, M, gm);
    else
      hipLaunchKernelGGL((linear_gemm_kernel<true, false, false>), grid,
                         dim3(256), 0, stream, dy, x, nullptr, dw, C, M, B, C, M, gm);
  }
  if (db) {
    hipLaunchKernelGGL(colsum_kernel, dim3((C + 255) / 256), dim3(256), 0,
                       stream, dy, db, B, C);
  }
}
