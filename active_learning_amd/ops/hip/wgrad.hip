// Convolution backward-weight (wgrad) on MFMA, NHWC bf16 -> fp32 dW.
//
//   dW[k, rsc] = sum_m dY[m, k] * im2col(x)[m, rsc],   m = (n,p,q) pixels
//
// Both operands have the contraction (m) as their SLOW dim, so tiles are
// staged into LDS TRANSPOSED ([channel][m], +8-element pad keeping rows
// 16B-aligned) and fragments then load as contiguous-m 16B ds_read_b128.
// Staging is double-buffered with the async-STAGE split (guide T14 /
// Guideline 15): the next m-tile's global loads are issued into registers
// BEFORE the current tile's MFMAs, and the transposed ds_write happens after
// the barrier — HBM latency hides under the MFMA phase.
// Split-K over m with fp32 atomicAdd into the dW accumulator (zeroed by the
// caller); fp32 output feeds the master-weight update exactly.
//
// Tile: 64(k) x 64(rsc) x 64(m), 4 waves (2x2), 32x32 per wave. Shapes with
// C % 8 != 0 or K % 8 != 0 (the stems) take the packed-im2col path in
// ops/functional.py, so the direct fallback below is a safety net only.

#include "al_common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

struct WgradShape {
  int N, H, W, C, K, R, S, P, Q, stride, pad;
  long L;   // contraction length N*P*Q
  int Nw;   // R*S*C
};

AL_DEV bool x_chunk(const bf16* __restrict__ x, const WgradShape& sh, long m, int rsc,
                    const bf16** out) {
  if (m >= sh.L || rsc >= sh.Nw) return false;
  const int q = (int)(m % sh.Q);
  long t = m / sh.Q;
  const int p = (int)(t % sh.P);
  const int n = (int)(t / sh.P);
  const int c = rsc % sh.C;
  const int rs = rsc / sh.C;
  const int s = rs % sh.S;
  const int r = rs / sh.S;
  const int h = p * sh.stride + r - sh.pad;
  const int w = q * sh.stride + s - sh.pad;
  if (h < 0 || h >= sh.H || w < 0 || w >= sh.W) return false;
  *out = x + (((long)n * sh.H + h) * sh.W + w) * sh.C + c;
  return true;
}

__launch_bounds__(256)
__global__ void wgrad_kernel(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                             float* __restrict__ dw, WgradShape sh, int grid_k,
                             long l_per_z) {
  constexpr int BMK = 64, BNW = 64, BL = 64, PAD = 8, LDT = BL + PAD;  // 72
  const int bk = blockIdx.x % grid_k;
  const int bn = blockIdx.x / grid_k;
  const int k0 = bk * BMK;
  const int n0 = bn * BNW;
  const long lz0 = (long)blockIdx.y * l_per_z;
  const long lz1 = min(sh.L, lz0 + l_per_z);
  if (lz0 >= lz1) return;

  __shared__ __attribute__((aligned(16))) bf16 At[2][BMK][LDT];  // [k][m]
  __shared__ __attribute__((aligned(16))) bf16 Bt[2][BNW][LDT];  // [rsc][m]

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wr = wid >> 1, wc = wid & 1;
  const int l15 = lane & 15, l4 = lane >> 4;

  // each thread owns two 8-deep channel chunks per operand per tile:
  //   chunk t in [0,512): ml = t>>3 (m within tile), u = t&7 (channel chunk)
  const int ml0 = tid >> 3, u0 = tid & 7;           // chunk tid
  const int ml1 = (tid + 256) >> 3, u1 = tid & 7;   // chunk tid+256

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  auto load_tile = [&](long m0, s16x8 va[2], s16x8 vb[2]) {
    const int mls[2] = {ml0, ml1};
    const int us[2] = {u0, u1};
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const long m = m0 + mls[i];
      va[i] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      vb[i] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      if (m < lz1) {
        if (k0 + us[i] * 8 < sh.K)
          va[i] = *(const s16x8*)(dy + m * sh.K + k0 + us[i] * 8);
        const bf16* src;
        if (x_chunk(x, sh, m, n0 + us[i] * 8, &src)) vb[i] = *(const s16x8*)src;
      }
    }
  };

  auto write_tile = [&](int buf, const s16x8 va[2], const s16x8 vb[2]) {
    const int mls[2] = {ml0, ml1};
    const int us[2] = {u0, u1};
#pragma unroll
    for (int i = 0; i < 2; ++i) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        short aj = va[i][j], bj = vb[i][j];
        At[buf][us[i] * 8 + j][mls[i]] = *(bf16*)&aj;
        Bt[buf][us[i] * 8 + j][mls[i]] = *(bf16*)&bj;
      }
    }
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {
      bf16x8 afrag[2], bfrag[2];
#pragma unroll
      for (int f = 0; f < 2; ++f) {
        afrag[f] = *(const bf16x8*)(&At[buf][wr * 32 + f * 16 + l15][mc * 32 + l4 * 8]);
        bfrag[f] = *(const bf16x8*)(&Bt[buf][wc * 32 + f * 16 + l15][mc * 32 + l4 * 8]);
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  s16x8 va[2], vb[2];
  load_tile(lz0, va, vb);
  write_tile(0, va, vb);
  int buf = 0;
  for (long m0 = lz0; m0 < lz1; m0 += BL) {
    __syncthreads();
    if (m0 + BL < lz1) load_tile(m0 + BL, va, vb);  // overlap with MFMAs below
    compute(buf);
    __syncthreads();
    if (m0 + BL < lz1) write_tile(buf ^ 1, va, vb);
    buf ^= 1;
  }

  // accumulate into global dW (fp32): D row = k, col = rsc
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int col = n0 + wc * 32 + ni * 16 + l15;
      if (col >= sh.Nw) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = k0 + wr * 32 + mi * 16 + l4 * 4 + r;
        if (row < sh.K) atomicAdd(&dw[(long)row * sh.Nw + col], acc[mi][ni][r]);
      }
    }
  }
}

// direct fallback: one dW element per thread, strided over L (safety net;
// normal stems go through the packed-im2col MFMA path)
__global__ void wgrad_direct_kernel(const bf16* __restrict__ dy,
                                    const bf16* __restrict__ x,
                                    float* __restrict__ dw, WgradShape sh) {
  const long total = (long)sh.K * sh.Nw;
  const long lz = (sh.L + gridDim.y - 1) / gridDim.y;
  const long m0 = (long)blockIdx.y * lz;
  const long m1 = min(sh.L, m0 + lz);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int k = (int)(i / sh.Nw);
    const int rsc = (int)(i % sh.Nw);
    float acc = 0.f;
    for (long m = m0; m < m1; ++m) {
      const bf16* px;
      if (!x_chunk(x, sh, m, rsc, &px)) continue;
      acc += bf2f(dy[m * sh.K + k]) * bf2f(*px);
    }
    if (gridDim.y == 1) dw[i] = acc;
    else atomicAdd(&dw[i], acc);
  }
}

extern "C" void al_conv2d_wgrad(const void* dy, const void* x, float* dw, int N, int H,
                                int W, int C, int K, int R, int S, int P, int Q,
                                int stride, int pad, hipStream_t stream) {
  WgradShape sh;
  sh.N = N; sh.H = H; sh.W = W; sh.C = C; sh.K = K; sh.R = R; sh.S = S;
  sh.P = P; sh.Q = Q; sh.stride = stride; sh.pad = pad;
  sh.L = (long)N * P * Q;
  sh.Nw = R * S * C;
  if (C % 8 == 0 && K % 8 == 0) {
    const int grid_k = (K + 63) / 64;
    const int grid_n = (sh.Nw + 63) / 64;
    const int tiles = grid_k * grid_n;
    // split-K: aim for >= 512 blocks to fill 256 CUs
    int z = (int)min((long)128, max((long)1, (512L + tiles - 1) / tiles));
    z = (int)min((long)z, max((long)1, sh.L / 64));
    long l_per_z = (sh.L + z - 1) / z;
    l_per_z = ((l_per_z + 63) / 64) * 64;
    z = (int)((sh.L + l_per_z - 1) / l_per_z);
    dim3 grid(tiles, z), block(256);
    hipLaunchKernelGGL(wgrad_kernel, grid, block, 0, stream, (const bf16*)dy,
                       (const bf16*)x, dw, sh, grid_k, l_per_z);
  } else {
    long total = (long)K * sh.Nw;
    int bx = (int)min((total + 255) / 256, (long)1024);
    int z = (int)min((long)64, max((long)1, sh.L / 8192));
    hipLaunchKernelGGL(wgrad_direct_kernel, dim3(bx, z), dim3(256), 0, stream,
                       (const bf16*)dy, (const bf16*)x, dw, sh);
  }
}
