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
#include <stdlib.h>

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

// ---------------------------------------------------------------------------
// 8x8 in-register bf16 transpose across a contiguous 8-lane group.
// Input:  lane l (within its group) holds row m = (l&7): 8 channel values
// Output: lane l holds channel (l&7): 8 m values
// 3 butterfly stages exchanging lane-bit k with element-bit k; the
// element-bit-0 stage mixes halves of dwords via v_perm_b32.
// ---------------------------------------------------------------------------
typedef unsigned int u32;

AL_DEV void xpose8x8(u32 d[4], int lane) {
  // stage 1: lane bit0 <-> element bit0 (within-dword halves)
  {
    u32 x[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) x[k] = __shfl_xor((int)d[k], 1, 64);
    const bool hi = lane & 1;
#pragma unroll
    for (int k = 0; k < 4; ++k)
      d[k] = hi ? __builtin_amdgcn_perm(d[k], x[k], 0x07060302)
                : __builtin_amdgcn_perm(x[k], d[k], 0x05040100);
  }
  // stage 2: lane bit1 <-> element bit1 (dword pairs 0<->1, 2<->3)
  {
    u32 x[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) x[k] = __shfl_xor((int)d[k], 2, 64);
    const bool hi = lane & 2;
    u32 n0 = hi ? x[1] : d[0];
    u32 n1 = hi ? d[1] : x[0];
    u32 n2 = hi ? x[3] : d[2];
    u32 n3 = hi ? d[3] : x[2];
    d[0] = n0; d[1] = n1; d[2] = n2; d[3] = n3;
  }
  // stage 3: lane bit2 <-> element bit2 (dword pairs 0<->2, 1<->3)
  {
    u32 x[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) x[k] = __shfl_xor((int)d[k], 4, 64);
    const bool hi = lane & 4;
    u32 n0 = hi ? x[2] : d[0];
    u32 n1 = hi ? x[3] : d[1];
    u32 n2 = hi ? d[2] : x[0];
    u32 n3 = hi ? d[3] : x[1];
    d[0] = n0; d[1] = n1; d[2] = n2; d[3] = n3;
  }
}

// GKR x GNC wave grid (4 waves, 64x64 tiles): (2,2) -> 128(k) x 128(rsc);
// (1,4) -> 64 x 256 for K=64 layers.
template <int GKR, int GNC>
__launch_bounds__(256)
__global__ void wgrad_kernel(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                             float* __restrict__ dw, WgradShape sh, int grid_k,
                             long l_per_z) {
  constexpr int BMK = GKR * 64, BNW = GNC * 64, BL = 64;
  constexpr int NAB = 2 * GKR, NBB = 2 * GNC, HA = NAB / 2, HB = NBB / 2;
  const int bk = blockIdx.x % grid_k;
  const int bn = blockIdx.x / grid_k;
  const int k0 = bk * BMK;
  const int n0 = bn * BNW;
  const long lz0 = (long)blockIdx.y * l_per_z;
  const long lz1 = min(sh.L, lz0 + l_per_z);
  if (lz0 >= lz1) return;

  // [ch][m] tiles, XOR-swizzled 16B slots (slot ^= ch&7): conflict-free
  // ds_write_b128 after the register transpose, igemm-style ds_read_b128.
  __shared__ __attribute__((aligned(16))) bf16 At[2][BMK][BL];
  __shared__ __attribute__((aligned(16))) bf16 Bt[2][BNW][BL];

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wr = wid / GNC, wc = wid % GNC;
  const int l15 = lane & 15, l4 = lane >> 4;
  const int u = lane >> 3;       // channel chunk (8 ch)
  const int mo = lane & 7;       // m offset within the wave's 8-m slice

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // sub-batches of (8-m slice x 8-ch chunk): A covers 64 m x BMK channels
  // in NAB batches, B covers 64 m x BNW channels in NBB batches.
  auto load_tile = [&](long m0, s16x8 va[NAB], s16x8 vb[NBB]) {
#pragma unroll
    for (int i = 0; i < NAB; ++i) {
      const long m = m0 + wid * 8 + 32 * (i / HA) + mo;
      const int uc = u + 8 * (i % HA);
      va[i] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      if (m < lz1 && k0 + uc * 8 < sh.K)
        va[i] = *(const s16x8*)(dy + m * sh.K + k0 + uc * 8);
    }
#pragma unroll
    for (int i = 0; i < NBB; ++i) {
      const long m = m0 + wid * 8 + 32 * (i / HB) + mo;
      const int uc = u + 8 * (i % HB);
      vb[i] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      const bf16* src;
      if (m < lz1 && x_chunk(x, sh, m, n0 + uc * 8, &src))
        vb[i] = *(const s16x8*)src;
    }
  };

  auto write_tile = [&](int buf, s16x8 va[NAB], s16x8 vb[NBB]) {
#pragma unroll
    for (int i = 0; i < NAB; ++i) {
      const int slot = wid + 4 * (i / HA);             // m-slot (16B = 8 m)
      const int ch = (u + 8 * (i % HA)) * 8 + mo;      // ch row
      const int sw = slot ^ (ch & 7);
      xpose8x8((u32*)&va[i], lane);
      *(s16x8*)(&At[buf][ch][sw * 8]) = va[i];
    }
#pragma unroll
    for (int i = 0; i < NBB; ++i) {
      const int slot = wid + 4 * (i / HB);
      const int ch = (u + 8 * (i % HB)) * 8 + mo;
      const int sw = slot ^ (ch & 7);
      xpose8x8((u32*)&vb[i], lane);
      *(s16x8*)(&Bt[buf][ch][sw * 8]) = vb[i];
    }
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int ar = wr * 64 + f * 16 + l15;
        const int as = (mc * 4 + l4) ^ (ar & 7);
        afrag[f] = *(const bf16x8*)(&At[buf][ar][as * 8]);
        const int br = wc * 64 + f * 16 + l15;
        const int bs = (mc * 4 + l4) ^ (br & 7);
        bfrag[f] = *(const bf16x8*)(&Bt[buf][br][bs * 8]);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  s16x8 va[NAB], vb[NBB];
  load_tile(lz0, va, vb);
  write_tile(0, va, vb);
  int buf = 0;
  for (long m0 = lz0; m0 < lz1; m0 += BL) {
    __syncthreads();  // write(buf) visible to all readers of this tile
    if (m0 + BL < lz1) load_tile(m0 + BL, va, vb);  // overlap with MFMAs below
    compute(buf);
    // no barrier: write targets buf^1, last read two phases ago
    if (m0 + BL < lz1) write_tile(buf ^ 1, va, vb);
    buf ^= 1;
  }

  // accumulate into global dW (fp32): D row = k, col = rsc
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int col = n0 + wc * 64 + ni * 16 + l15;
      if (col >= sh.Nw) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = k0 + wr * 64 + mi * 16 + l4 * 4 + r;
        if (row < sh.K) atomicAdd(&dw[(long)row * sh.Nw + col], acc[mi][ni][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad v2: row-major staging + hardware transpose reads.
//
// v1 stages [ch][m] tiles via an in-register 8x8 bf16 transpose (3 shfl
// butterflies + v_perm per chunk) so fragments can ds_read_b128. gfx950's
// ds_read_b64_tr_b16 makes that transpose free: stage both operands ROW-major
// ([l][ch] — every 16B global gather is contiguous, issued as
// global_load_lds with no VGPR round trip or ds_write pass) and read MFMA
// fragments with the transpose load. Probe-verified semantics
// (tools/tr_probe.hip): within a [4][16]-u16 128B-aligned block, lane l
// receives rows 4*(l>>4)+0..3 of column l&15 — exactly the A/B fragment
// gather (lane's own 8B-aligned address selects block and column).
//
// LDS per operand per buffer: [width/16 groups][64 l][16 ch] bf16 (row-major
// within each 16-channel group — the conflict-free tr_b16 subtiling).
// dy chunks advance along L with a pointer bump; x chunks keep per-slot
// (n,p,q) pixel state advanced by precomputed (dn,dp,dq) carries.
// ---------------------------------------------------------------------------

template <int GKR, int GNC, bool SCAL = false>
__launch_bounds__(256)
__global__ void wgrad2_kernel(const bf16* __restrict__ dy,
                              const bf16* __restrict__ x,
                              float* __restrict__ dw,
                              const bf16* __restrict__ zero,
                              WgradShape sh, int grid_k, long l_per_z) {
  constexpr int BMK = GKR * 64, BNW = GNC * 64, BL = 64;
  constexpr int ACH = BMK * BL;            // elements per A buffer
  constexpr int BCH = BNW * BL;
  constexpr int NA = ACH / (256 * 8);      // A 16B chunks per thread
  constexpr int NB = BCH / (256 * 8);
  const int bk = blockIdx.x % grid_k;
  const int bn = blockIdx.x / grid_k;
  const int k0 = bk * BMK;
  const int n0 = bn * BNW;
  const long lz0 = (long)blockIdx.y * l_per_z;
  const long lz1 = min(sh.L, lz0 + l_per_z);
  if (lz0 >= lz1) return;

  __shared__ __attribute__((aligned(16))) bf16 As2[2][ACH];
  __shared__ __attribute__((aligned(16))) bf16 Bs2[2][BCH];

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wr = wid / GNC, wc = wid % GNC;
  const int l15 = lane & 15, l4 = lane >> 4;

  // ---- per-slot staging state. Chunk ci = slot*256 + tid maps to
  // (group g, l-row, half h): g = ci>>7, l = (ci&127)>>1, h = ci&1; dest
  // element ci*8 (lane-linear for global_load_lds).
  // A (dy): source = dy + (lz0+l)*K + k0 + g*16 + h*8 — pointer bump only.
  long a_off[NA];
  int a_l[NA];
  bool a_chok[NA];
#pragma unroll
  for (int i = 0; i < NA; ++i) {
    const int ci = i * 256 + tid;
    const int g = ci >> 7, l = (ci & 127) >> 1, h = ci & 1;
    const int ch = k0 + g * 16 + h * 8;
    a_l[i] = l;
    a_chok[i] = ch + 8 <= sh.K;
    a_off[i] = (lz0 + l) * (long)sh.K + ch;
  }
  // B (x gather): nw = n0 + g*16 + h*8 -> fixed (r,s,c); pixel (n,p,q) from
  // l = lz0 + l_fix, advanced by BL each step with precomputed carries.
  int b_n[NB], b_p[NB], b_q[NB], b_ho[NB], b_wo[NB];
  long b_coff[NB];
  int b_l[NB];
  bool b_chok[NB];
  const int dn = (int)(BL / ((long)sh.P * sh.Q));
  const int rem = (int)(BL % ((long)sh.P * sh.Q));
  const int dp = rem / sh.Q, dq = rem % sh.Q;
#pragma unroll
  for (int i = 0; i < NB; ++i) {
    const int ci = i * 256 + tid;
    const int g = ci >> 7, l = (ci & 127) >> 1, h = ci & 1;
    const int nw = n0 + g * 16 + h * 8;
    b_l[i] = l;
    b_chok[i] = nw + 8 <= sh.Nw;
    const int c = nw % sh.C;
    const int rs = nw / sh.C;
    b_ho[i] = (rs / sh.S) - sh.pad;         // r - pad
    b_wo[i] = (rs % sh.S) - sh.pad;         // s - pad
    b_coff[i] = c;
    const long m = lz0 + l;
    b_q[i] = (int)(m % sh.Q);
    long t = m / sh.Q;
    b_p[i] = (int)(t % sh.P);
    b_n[i] = (int)(t / sh.P);
  }
  auto b_advance = [&]() {
#pragma unroll
    for (int i = 0; i < NB; ++i) {
      b_q[i] += dq;
      if (b_q[i] >= sh.Q) { b_q[i] -= sh.Q; ++b_p[i]; }
      b_p[i] += dp;
      if (b_p[i] >= sh.P) { b_p[i] -= sh.P; ++b_n[i]; }
      b_n[i] += dn;
    }
  };

  auto stage = [&](int buf, long l0) {
#pragma unroll
    for (int i = 0; i < NA; ++i) {
      const bf16* src = (a_chok[i] && l0 + a_l[i] < lz1) ? dy + a_off[i] : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(&As2[buf][(i * 256 + wid * 64) * 8]),
          16, 0, 0);
      a_off[i] += (long)BL * sh.K;
    }
#pragma unroll
    for (int i = 0; i < NB; ++i) {
      const int hh = b_p[i] * sh.stride + b_ho[i];
      const int ww = b_q[i] * sh.stride + b_wo[i];
      const bool ok = b_chok[i] && (l0 + b_l[i] < lz1) &&
                      (unsigned)hh < (unsigned)sh.H && (unsigned)ww < (unsigned)sh.W;
      const bf16* src = ok
          ? x + (((long)b_n[i] * sh.H + hh) * sh.W + ww) * sh.C + b_coff[i]
          : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(&Bs2[buf][(i * 256 + wid * 64) * 8]),
          16, 0, 0);
    }
    b_advance();
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // tr_b16 fragment read: two b64 transpose loads give lane l column l&15,
  // l-rows (l>>4)*8 + 0..7 of a [64][16] group image — the MFMA operand.
  static_assert(true, "");
  auto scalfrag = [&](const bf16* img, int group, int mc) -> bf16x8 {
    // diagnostic fallback (AL_WGRAD_SCAL=1): same gather via compiler-
    // managed scalar LDS reads
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = img[group * 1024 + (mc * 32 + l4 * 8 + j) * 16 + l15];
    return o;
  };
  typedef __attribute__((address_space(3))) s16x4* lds_v4p;
  auto trfrag = [&](const bf16* img, int group, int mc) -> bf16x8 {
    // compiler-modeled transpose loads (proper lgkmcnt tracking and
    // scheduling — a hand-rolled asm version suffered occupancy-dependent
    // reordering hazards)
    const bf16* p = img + group * 1024 + mc * 32 * 16 + l4 * 8 * 16 + l15 * 4;
    union { s16x4 h[2]; bf16x8 v; } u;
    u.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_v4p)p);
    u.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_v4p)(p + 64));
    return u.v;
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {
      bf16x8 afrag[4], bfrag[4];
      if (SCAL) {
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          afrag[f] = scalfrag(As2[buf], wr * 4 + f, mc);
          bfrag[f] = scalfrag(Bs2[buf], wc * 4 + f, mc);
        }
      } else {
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          afrag[f] = trfrag(As2[buf], wr * 4 + f, mc);
          bfrag[f] = trfrag(Bs2[buf], wc * 4 + f, mc);
        }
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  stage(0, lz0);
  int buf = 0;
  for (long m0 = lz0; m0 < lz1; m0 += BL) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (m0 + BL < lz1) stage(buf ^ 1, m0 + BL);
    compute(buf);
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int col = n0 + wc * 64 + ni * 16 + l15;
      if (col >= sh.Nw) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = k0 + wr * 64 + mi * 16 + l4 * 4 + r;
        if (row < sh.K) atomicAdd(&dw[(long)row * sh.Nw + col], acc[mi][ni][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad v3 (experimental, AL_WGRAD_V3): same row-major + tr_b16 structure as
// v2 but 512 threads and a 3-buffer ring with counted vmcnt — the staging
// for step s+2 is issued before step s's MFMAs and only vmcnt(4) (one stage
// in flight) is waited at each step top, so no vmcnt(0) drain ever happens
// in the loop. 96 KiB LDS -> 1 block/CU but still 8 waves (2/SIMD).
// ---------------------------------------------------------------------------

template <int GKR, int GNC>
__launch_bounds__(512)
__global__ void wgrad3_kernel(const bf16* __restrict__ dy,
                              const bf16* __restrict__ x,
                              float* __restrict__ dw,
                              const bf16* __restrict__ zero,
                              WgradShape sh, int grid_k, long l_per_z) {
  constexpr int BMK = GKR * 64, BNW = GNC * 64, BL = 64;
  constexpr int ACH = BMK * BL;
  constexpr int BCH = BNW * BL;
  constexpr int NA = ACH / (512 * 8);      // 16B chunks per thread
  constexpr int NB = BCH / (512 * 8);
  const int bk = blockIdx.x % grid_k;
  const int bn = blockIdx.x / grid_k;
  const int k0 = bk * BMK;
  const int n0 = bn * BNW;
  const long lz0 = (long)blockIdx.y * l_per_z;
  const long lz1 = min(sh.L, lz0 + l_per_z);
  if (lz0 >= lz1) return;

  __shared__ __attribute__((aligned(16))) bf16 As3[3][ACH];
  __shared__ __attribute__((aligned(16))) bf16 Bs3[3][BCH];

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wr = wid >> 2, wc = wid & 3;   // 2x4 wave grid, 64x32 each
  const int l15 = lane & 15, l4 = lane >> 4;

  long a_off[NA];
  int a_l[NA];
  bool a_chok[NA];
#pragma unroll
  for (int i = 0; i < NA; ++i) {
    const int ci = i * 512 + tid;
    const int g = ci >> 7, l = (ci & 127) >> 1, h = ci & 1;
    const int ch = k0 + g * 16 + h * 8;
    a_l[i] = l;
    a_chok[i] = ch + 8 <= sh.K;
    a_off[i] = (lz0 + l) * (long)sh.K + ch;
  }
  int b_n[NB], b_p[NB], b_q[NB], b_ho[NB], b_wo[NB];
  long b_coff[NB];
  int b_l[NB];
  bool b_chok[NB];
  const int dn3 = (int)(BL / ((long)sh.P * sh.Q));
  const int rem3 = (int)(BL % ((long)sh.P * sh.Q));
  const int dp3 = rem3 / sh.Q, dq3 = rem3 % sh.Q;
#pragma unroll
  for (int i = 0; i < NB; ++i) {
    const int ci = i * 512 + tid;
    const int g = ci >> 7, l = (ci & 127) >> 1, h = ci & 1;
    const int nw = n0 + g * 16 + h * 8;
    b_l[i] = l;
    b_chok[i] = nw + 8 <= sh.Nw;
    const int c = nw % sh.C;
    const int rs = nw / sh.C;
    b_ho[i] = (rs / sh.S) - sh.pad;
    b_wo[i] = (rs % sh.S) - sh.pad;
    b_coff[i] = c;
    const long m = lz0 + l;
    b_q[i] = (int)(m % sh.Q);
    long t = m / sh.Q;
    b_p[i] = (int)(t % sh.P);
    b_n[i] = (int)(t / sh.P);
  }

  auto stage = [&](int buf, long l0, bool live) {
#pragma unroll
    for (int i = 0; i < NA; ++i) {
      const bf16* src = (live && a_chok[i] && l0 + a_l[i] < lz1)
                            ? dy + a_off[i] : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(&As3[buf][(i * 512 + wid * 64) * 8]),
          16, 0, 0);
      a_off[i] += (long)BL * sh.K;
    }
#pragma unroll
    for (int i = 0; i < NB; ++i) {
      const int hh = b_p[i] * sh.stride + b_ho[i];
      const int ww = b_q[i] * sh.stride + b_wo[i];
      const bool ok = live && b_chok[i] && (l0 + b_l[i] < lz1) &&
                      (unsigned)hh < (unsigned)sh.H && (unsigned)ww < (unsigned)sh.W;
      const bf16* src = ok
          ? x + (((long)b_n[i] * sh.H + hh) * sh.W + ww) * sh.C + b_coff[i]
          : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(&Bs3[buf][(i * 512 + wid * 64) * 8]),
          16, 0, 0);
      b_q[i] += dq3;
      if (b_q[i] >= sh.Q) { b_q[i] -= sh.Q; ++b_p[i]; }
      b_p[i] += dp3;
      if (b_p[i] >= sh.P) { b_p[i] -= sh.P; ++b_n[i]; }
      b_n[i] += dn3;
    }
  };

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  typedef __attribute__((address_space(3))) s16x4* lds3_v4p;
  auto trfrag = [&](const bf16* img, int group, int mc) -> bf16x8 {
    const bf16* p = img + group * 1024 + mc * 32 * 16 + l4 * 8 * 16 + l15 * 4;
    union { s16x4 h[2]; bf16x8 v; } u;
    u.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds3_v4p)p);
    u.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds3_v4p)(p + 64));
    return u.v;
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {
      bf16x8 afrag[4], bfrag[2];
#pragma unroll
      for (int f = 0; f < 4; ++f) afrag[f] = trfrag(As3[buf], wr * 4 + f, mc);
#pragma unroll
      for (int f = 0; f < 2; ++f) bfrag[f] = trfrag(Bs3[buf], wc * 2 + f, mc);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  const long steps = (lz1 - lz0 + BL - 1) / BL;
  stage(0, lz0, true);
  stage(1, lz0 + BL, 1 < steps);
  for (long s = 0; s < steps; ++s) {
    // counted wait: only the NEXT stage's 4 loads may stay in flight
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __syncthreads();
    stage((int)((s + 2) % 3), lz0 + (s + 2) * BL, s + 2 < steps);
    compute((int)(s % 3));
  }

#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int col = n0 + wc * 32 + ni * 16 + l15;
      if (col >= sh.Nw) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = k0 + wr * 64 + mi * 16 + l4 * 4 + r;
        if (row < sh.K) atomicAdd(&dw[(long)row * sh.Nw + col], acc[mi][ni][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad v4: the 8-phase 256² deep-pipelined schedule (guide §5 template)
// applied to backward-weight. 256(k) x 256(rsc) tile, BL=64 contraction
// steps over m, 8 waves (512 threads) in a 2x4 grid, 128 KiB LDS as a ring
// of 4 half-tiles per operand, one half refilled per phase, vmcnt(4) once
// per m-tile group — no vmcnt(0) drain in the loop (the 2-buffer v2 drains
// every 64-m step, which is its measured ceiling: 235 TF vs the template's
// ~1330, profiles/pmc_counters.md).
//
// Staging and fragment reads are v2's: both operands stage ROW-major
// ([m-l][16ch] groups, every global gather a contiguous 16 B
// global_load_lds) and MFMA fragments come out via ds_read_b64_tr_b16
// hardware-transpose reads. dy chunks advance along m with a pointer bump;
// x chunks keep per-slot (n,p,q) pixel state with precomputed carries, one
// state per (half, li) slot since halves refill on different phases.
// ---------------------------------------------------------------------------

__launch_bounds__(512, 1)
__global__ void wgrad4_kernel(const bf16* __restrict__ dy,
                              const bf16* __restrict__ x,
                              float* __restrict__ dw,
                              const bf16* __restrict__ zero,
                              WgradShape sh, int grid_k, long l_per_z) {
  constexpr int BMK = 256, BNW = 256, BL = 64;
  constexpr int HALF = 128 * BL;       // elements per half-tile (A and B)
  constexpr int NL = 2;                // 16B chunks per thread per half refill
  const int bk = blockIdx.x % grid_k;
  const int bn = blockIdx.x / grid_k;
  const int k0 = bk * BMK;
  const int n0 = bn * BNW;
  const long lz0 = (long)blockIdx.y * l_per_z;
  const long lz1 = min(sh.L, lz0 + l_per_z);
  if (lz0 >= lz1) return;
  const int KT = (int)((lz1 - lz0 + BL - 1) / BL);

  extern __shared__ __attribute__((aligned(16))) char smem4[];
  bf16* As = (bf16*)smem4;             // 4 x HALF (slot = buf*2 + half)
  bf16* Bs = As + 4 * HALF;

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wrq = wid >> 2, wcq = wid & 3;
  const int l15 = lane & 15, l4 = lane >> 4;

  // ---- A (dy) slot state: 2 halves x NL chunks
  long a_off[2][NL];
  bool a_chok[2][NL];
  int a_l[2][NL];
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      const int ci = li * 512 + tid;
      const int g = ci >> 7, l = (ci & 127) >> 1, h2 = ci & 1;
      const int ch = k0 + h * 128 + g * 16 + h2 * 8;
      a_l[h][li] = l;
      a_chok[h][li] = ch + 8 <= sh.K;
      a_off[h][li] = (lz0 + l) * (long)sh.K + ch;
    }
  // ---- B (x) slot state
  int b_n[2][NL], b_p[2][NL], b_q[2][NL], b_ho[2][NL], b_wo[2][NL], b_l[2][NL];
  long b_coff[2][NL];
  bool b_chok[2][NL];
  const int dn = (int)(BL / ((long)sh.P * sh.Q));
  const int rem = (int)(BL % ((long)sh.P * sh.Q));
  const int dp = rem / sh.Q, dq = rem % sh.Q;
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      const int ci = li * 512 + tid;
      const int g = ci >> 7, l = (ci & 127) >> 1, h2 = ci & 1;
      const int nw = n0 + h * 128 + g * 16 + h2 * 8;
      b_l[h][li] = l;
      b_chok[h][li] = nw + 8 <= sh.Nw;
      const int c = nw % sh.C;
      const int rs = nw / sh.C;
      b_ho[h][li] = (rs / sh.S) - sh.pad;
      b_wo[h][li] = (rs % sh.S) - sh.pad;
      b_coff[h][li] = c;
      const long m = lz0 + l;
      b_q[h][li] = (int)(m % sh.Q);
      long t = m / sh.Q;
      b_p[h][li] = (int)(t % sh.P);
      b_n[h][li] = (int)(t / sh.P);
    }
  auto advance_a = [&](int h) {
#pragma unroll
    for (int li = 0; li < NL; ++li) a_off[h][li] += (long)BL * sh.K;
  };
  auto advance_b = [&](int h) {
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      b_q[h][li] += dq;
      if (b_q[h][li] >= sh.Q) { b_q[h][li] -= sh.Q; ++b_p[h][li]; }
      b_p[h][li] += dp;
      if (b_p[h][li] >= sh.P) { b_p[h][li] -= sh.P; ++b_n[h][li]; }
      b_n[h][li] += dn;
    }
  };
  auto stage_a = [&](int h, int kt, bf16* slot) {
    const long l0 = lz0 + (long)kt * BL;
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      const bf16* src = (kt < KT && a_chok[h][li] && l0 + a_l[h][li] < lz1)
                            ? dy + a_off[h][li] : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(slot + (li * 512 + wid * 64) * 8),
          16, 0, 0);
    }
  };
  auto stage_b = [&](int h, int kt, bf16* slot) {
    const long l0 = lz0 + (long)kt * BL;
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      const int hh = b_p[h][li] * sh.stride + b_ho[h][li];
      const int ww = b_q[h][li] * sh.stride + b_wo[h][li];
      const bool ok = kt < KT && b_chok[h][li] && (l0 + b_l[h][li] < lz1) &&
                      (unsigned)hh < (unsigned)sh.H && (unsigned)ww < (unsigned)sh.W;
      const bf16* src = ok
          ? x + (((long)b_n[h][li] * sh.H + hh) * sh.W + ww) * sh.C + b_coff[h][li]
          : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(slot + (li * 512 + wid * 64) * 8),
          16, 0, 0);
    }
  };
  // refill schedule (as the 256² conv GEMM): group g consumes buf = g&1;
  // ph0: A h1 <- tile g+1, ph1: B h1 <- g+1, ph2: A h0 <- g+2, ph3: B h0 <- g+2
  auto refill = [&](int g, int ph) {
    const int buf = g & 1;
    switch (ph) {
      case 0: advance_a(1); stage_a(1, g + 1, As + ((buf ^ 1) * 2 + 1) * HALF); break;
      case 1: advance_b(1); stage_b(1, g + 1, Bs + ((buf ^ 1) * 2 + 1) * HALF); break;
      case 2: advance_a(0); stage_a(0, g + 2, As + (buf * 2) * HALF); break;
      case 3: advance_b(0); stage_b(0, g + 2, Bs + (buf * 2) * HALF); break;
    }
  };

  typedef __attribute__((address_space(3))) s16x4* lds4_v4p;
  auto trfrag = [&](const bf16* img, int group, int mc) -> bf16x8 {
    const bf16* p = img + group * 1024 + mc * 32 * 16 + l4 * 8 * 16 + l15 * 4;
    union { s16x4 h[2]; bf16x8 v; } u;
    u.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds4_v4p)p);
    u.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds4_v4p)(p + 64));
    return u.v;
  };

  f32x4 acc[2][2][4][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int f = 0; f < 4; ++f)
#pragma unroll
        for (int n = 0; n < 2; ++n) acc[i][j][f][n] = f32x4{0.f, 0.f, 0.f, 0.f};

  // prologue: tile 0 both halves + tile 1's h0 (6 half-stages = 12 loads in
  // flight per thread); h1 states stay at tile 0, h0 advanced to tile 1
  stage_a(0, 0, As);
  stage_b(0, 0, Bs);
  stage_a(1, 0, As + HALF);
  stage_b(1, 0, Bs + HALF);
  advance_a(0); advance_b(0);
  stage_a(0, 1, As + 2 * HALF);
  stage_b(0, 1, Bs + 2 * HALF);

  for (int g = 0; g < KT; ++g) {
    const int buf = g & 1;
    // drain + barrier: cross-wave retirement guarantee (see wgrad5_kernel)
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int qm = ph >> 1, qn = ph & 1;
      const bf16* ah = As + (buf * 2 + qm) * HALF;
      const bf16* bh = Bs + (buf * 2 + qn) * HALF;
      bf16x8 af[4][2], bf[2][2];
#pragma unroll
      for (int mc = 0; mc < 2; ++mc) {
#pragma unroll
        for (int fg = 0; fg < 4; ++fg) af[fg][mc] = trfrag(ah, wrq * 4 + fg, mc);
#pragma unroll
        for (int ng = 0; ng < 2; ++ng) bf[ng][mc] = trfrag(bh, wcq * 2 + ng, mc);
      }
      refill(g, ph);
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mc = 0; mc < 2; ++mc)
#pragma unroll
        for (int fg = 0; fg < 4; ++fg)
#pragma unroll
          for (int ng = 0; ng < 2; ++ng)
            acc[qm][qn][fg][ng] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[fg][mc], bf[ng][mc], acc[qm][qn][fg][ng], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // drain tail zero-page glds
#pragma unroll
  for (int qm = 0; qm < 2; ++qm)
#pragma unroll
    for (int qn = 0; qn < 2; ++qn)
#pragma unroll
      for (int fg = 0; fg < 4; ++fg)
#pragma unroll
        for (int ng = 0; ng < 2; ++ng) {
          const int col = n0 + qn * 128 + wcq * 32 + ng * 16 + l15;
          if (col >= sh.Nw) continue;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = k0 + qm * 128 + wrq * 64 + fg * 16 + l4 * 4 + r;
            if (row < sh.K)
              atomicAdd(&dw[(long)row * sh.Nw + col], acc[qm][qn][fg][ng][r]);
          }
        }
}

// ---------------------------------------------------------------------------
// wgrad v5: 2-phase variant of the 256² ring. v4's 4-phase schedule leaves
// only 8 MFMAs per wave per barrier interval (vs v2's 32 per drain) and
// measured BELOW v2 on most shapes. v5 processes a full A-half row of
// quadrants per phase — (qm,0) then (qm,1) with the B h1 fragments read
// mid-phase — giving 32 MFMAs per phase at the same register budget.
// B half-slots are consumed every phase, so their refills trail the phase's
// closing barrier; group-top wait is vmcnt(2) (only the trailing B h0
// prefetch may stay in flight).
// ---------------------------------------------------------------------------

__launch_bounds__(512, 1)
__global__ void wgrad5_kernel(const bf16* __restrict__ dy,
                              const bf16* __restrict__ x,
                              float* __restrict__ dw,
                              const bf16* __restrict__ zero,
                              WgradShape sh, int grid_k, long l_per_z) {
  constexpr int BL = 64;
  constexpr int HALF = 128 * BL;
  constexpr int NL = 2;
  const int bk = blockIdx.x % grid_k;
  const int bn = blockIdx.x / grid_k;
  const int k0 = bk * 256;
  const int n0 = bn * 256;
  const long lz0 = (long)blockIdx.y * l_per_z;
  const long lz1 = min(sh.L, lz0 + l_per_z);
  if (lz0 >= lz1) return;
  const int KT = (int)((lz1 - lz0 + BL - 1) / BL);

  extern __shared__ __attribute__((aligned(16))) char smem5[];
  bf16* As = (bf16*)smem5;
  bf16* Bs = As + 4 * HALF;

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wrq = wid >> 2, wcq = wid & 3;
  const int l15 = lane & 15, l4 = lane >> 4;

  long a_off[2][NL];
  bool a_chok[2][NL];
  int a_l[2][NL];
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      const int ci = li * 512 + tid;
      const int g = ci >> 7, l = (ci & 127) >> 1, h2 = ci & 1;
      const int ch = k0 + h * 128 + g * 16 + h2 * 8;
      a_l[h][li] = l;
      a_chok[h][li] = ch + 8 <= sh.K;
      a_off[h][li] = (lz0 + l) * (long)sh.K + ch;
    }
  int b_n[2][NL], b_p[2][NL], b_q[2][NL], b_ho[2][NL], b_wo[2][NL], b_l[2][NL];
  long b_coff[2][NL];
  bool b_chok[2][NL];
  const int dn = (int)(BL / ((long)sh.P * sh.Q));
  const int rem = (int)(BL % ((long)sh.P * sh.Q));
  const int dp = rem / sh.Q, dq = rem % sh.Q;
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      const int ci = li * 512 + tid;
      const int g = ci >> 7, l = (ci & 127) >> 1, h2 = ci & 1;
      const int nw = n0 + h * 128 + g * 16 + h2 * 8;
      b_l[h][li] = l;
      b_chok[h][li] = nw + 8 <= sh.Nw;
      const int c = nw % sh.C;
      const int rs = nw / sh.C;
      b_ho[h][li] = (rs / sh.S) - sh.pad;
      b_wo[h][li] = (rs % sh.S) - sh.pad;
      b_coff[h][li] = c;
      const long m = lz0 + l;
      b_q[h][li] = (int)(m % sh.Q);
      long t = m / sh.Q;
      b_p[h][li] = (int)(t % sh.P);
      b_n[h][li] = (int)(t / sh.P);
    }
  auto advance_a = [&](int h) {
#pragma unroll
    for (int li = 0; li < NL; ++li) a_off[h][li] += (long)BL * sh.K;
  };
  auto advance_b = [&](int h) {
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      b_q[h][li] += dq;
      if (b_q[h][li] >= sh.Q) { b_q[h][li] -= sh.Q; ++b_p[h][li]; }
      b_p[h][li] += dp;
      if (b_p[h][li] >= sh.P) { b_p[h][li] -= sh.P; ++b_n[h][li]; }
      b_n[h][li] += dn;
    }
  };
  auto stage_a = [&](int h, int kt, bf16* slot) {
    const long l0 = lz0 + (long)kt * BL;
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      const bf16* src = (kt < KT && a_chok[h][li] && l0 + a_l[h][li] < lz1)
                            ? dy + a_off[h][li] : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(slot + (li * 512 + wid * 64) * 8),
          16, 0, 0);
    }
  };
  auto stage_b = [&](int h, int kt, bf16* slot) {
    const long l0 = lz0 + (long)kt * BL;
#pragma unroll
    for (int li = 0; li < NL; ++li) {
      const int hh = b_p[h][li] * sh.stride + b_ho[h][li];
      const int ww = b_q[h][li] * sh.stride + b_wo[h][li];
      const bool ok = kt < KT && b_chok[h][li] && (l0 + b_l[h][li] < lz1) &&
                      (unsigned)hh < (unsigned)sh.H && (unsigned)ww < (unsigned)sh.W;
      const bf16* src = ok
          ? x + (((long)b_n[h][li] * sh.H + hh) * sh.W + ww) * sh.C + b_coff[h][li]
          : zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(slot + (li * 512 + wid * 64) * 8),
          16, 0, 0);
    }
  };

  typedef __attribute__((address_space(3))) s16x4* lds5_v4p;
  auto trfrag = [&](const bf16* img, int group, int mc) -> bf16x8 {
    const bf16* p = img + group * 1024 + mc * 32 * 16 + l4 * 8 * 16 + l15 * 4;
    union { s16x4 h[2]; bf16x8 v; } u;
    u.h[0] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds5_v4p)p);
    u.h[1] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds5_v4p)(p + 64));
    return u.v;
  };

  f32x4 acc[2][2][4][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int f = 0; f < 4; ++f)
#pragma unroll
        for (int n = 0; n < 2; ++n) acc[i][j][f][n] = f32x4{0.f, 0.f, 0.f, 0.f};

  // prologue (12 loads/thread): tile0 all four halves, then tile1's A h0 and
  // B h0 — matches the steady-state order so vmcnt(2) at the group top
  // drains everything except the trailing B h0 prefetch.
  stage_a(0, 0, As);
  stage_a(1, 0, As + HALF);
  stage_b(0, 0, Bs);
  stage_b(1, 0, Bs + HALF);
  advance_a(0); advance_b(0);
  stage_a(0, 1, As + 2 * HALF);
  stage_b(0, 1, Bs + 2 * HALF);

  for (int g = 0; g < KT; ++g) {
    const int buf = g & 1;
    // cross-wave handshake: EVERY wave drains to <=2 outstanding loads and
    // THEN barriers, so all waves' contributions to this group's slots are
    // retired before any wave reads them. Without the barrier the trailing
    // B refill (issued at the END of the previous group) has only ~1 phase
    // of slack and a fast wave reads a half-landed slot (race screen caught
    // intermittent nan/inf at KT=11 shapes).
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int qm = 0; qm < 2; ++qm) {
      const bf16* ah = As + (buf * 2 + qm) * HALF;
      const bf16* b0 = Bs + (buf * 2) * HALF;
      const bf16* b1 = Bs + (buf * 2 + 1) * HALF;
      bf16x8 af[4][2], bf[2][2];
#pragma unroll
      for (int mc = 0; mc < 2; ++mc) {
#pragma unroll
        for (int fg = 0; fg < 4; ++fg) af[fg][mc] = trfrag(ah, wrq * 4 + fg, mc);
#pragma unroll
        for (int ng = 0; ng < 2; ++ng) bf[ng][mc] = trfrag(b0, wcq * 2 + ng, mc);
      }
      if (qm == 0) { advance_a(1); stage_a(1, g + 1, As + ((buf ^ 1) * 2 + 1) * HALF); }
      else         { advance_a(0); stage_a(0, g + 2, As + (buf * 2) * HALF); }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mc = 0; mc < 2; ++mc)
#pragma unroll
        for (int fg = 0; fg < 4; ++fg)
#pragma unroll
          for (int ng = 0; ng < 2; ++ng)
            acc[qm][0][fg][ng] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[fg][mc], bf[ng][mc], acc[qm][0][fg][ng], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // second quadrant: B h1 fragments mid-phase (reads target the CURRENT
      // buf's B h1 slot; the in-flight refill writes a different slot)
#pragma unroll
      for (int mc = 0; mc < 2; ++mc)
#pragma unroll
        for (int ng = 0; ng < 2; ++ng) bf[ng][mc] = trfrag(b1, wcq * 2 + ng, mc);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mc = 0; mc < 2; ++mc)
#pragma unroll
        for (int fg = 0; fg < 4; ++fg)
#pragma unroll
          for (int ng = 0; ng < 2; ++ng)
            acc[qm][1][fg][ng] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[fg][mc], bf[ng][mc], acc[qm][1][fg][ng], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
    // trailing B refills: both B halves were last read inside this group,
    // and the closing barrier above orders every wave's reads before these
    // glds writes. B h1 <- g+1 first, then B h0 <- g+2 (the only loads
    // vmcnt(2) leaves in flight at the next group top).
    advance_b(1);
    stage_b(1, g + 1, Bs + ((buf ^ 1) * 2 + 1) * HALF);
    advance_b(0);
    stage_b(0, g + 2, Bs + (buf * 2) * HALF);
  }

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#pragma unroll
  for (int qm = 0; qm < 2; ++qm)
#pragma unroll
    for (int qn = 0; qn < 2; ++qn)
#pragma unroll
      for (int fg = 0; fg < 4; ++fg)
#pragma unroll
        for (int ng = 0; ng < 2; ++ng) {
          const int col = n0 + qn * 128 + wcq * 32 + ng * 16 + l15;
          if (col >= sh.Nw) continue;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = k0 + qm * 128 + wrq * 64 + fg * 16 + l4 * 4 + r;
            if (row < sh.K)
              atomicAdd(&dw[(long)row * sh.Nw + col], acc[qm][qn][fg][ng][r]);
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

extern "C" void al_conv2d_wgrad(const void* dy, const void* x, float* dw,
                                const void* zero_page, int N, int H,
                                int W, int C, int K, int R, int S, int P, int Q,
                                int stride, int pad, hipStream_t stream) {
  WgradShape sh;
  sh.N = N; sh.H = H; sh.W = W; sh.C = C; sh.K = K; sh.R = R; sh.S = S;
  sh.P = P; sh.Q = Q; sh.stride = stride; sh.pad = pad;
  sh.L = (long)N * P * Q;
  sh.Nw = R * S * C;
  if (C % 8 == 0 && K % 8 == 0) {
    // 256² ring schedules for the wide layers (AL_WGRAD_V4: 0=off, 1=v4
    // 4-phase, 2=v5 2-phase). Need a K-chain long enough to amortize the
    // 3-tile-deep prologue (KT >= threshold) and enough (k, rsc) coverage
    // that the 256² tile isn't mostly zero-page. Measured (tools/ab_wgrad):
    // v4 beats v2 only on l3.conv2-class shapes; v5 doubles the MFMA per
    // barrier interval.
    // Measured (tools/ab_wgrad, B=256 shapes): the 256² ring beats the
    // 2-barrier v2 only on the 3x3 K=256 class (l3.conv2: v5 0.137 ms vs v2
    // 0.154); everywhere else v2's 2-blocks/CU overlap wins (64 KiB LDS vs
    // the ring's 128 KiB -> 1 block/CU), the same occupancy regime that
    // killed the v3 ring. Default routing: v5 on that class only.
    static int v4 = -2, v4_minkt = 12;
    if (v4 == -2) {
      const char* e = getenv("AL_WGRAD_V4");
      v4 = e ? atoi(e) : 3;   // 0 off, 1 v4 all, 2 v5 all, 3 v5 where it wins
      const char* t = getenv("AL_WGRAD_V4_MINKT");
      if (t) v4_minkt = atoi(t);
    }
    const bool v5_wins = (R * S > 1) && K >= 192 && K < 384 && sh.Nw >= 2048;
    if ((v4 == 1 || v4 == 2 || (v4 == 3 && v5_wins)) && K >= 192 && sh.Nw >= 192) {
      const int grid_k4 = (K + 255) / 256;
      const int grid_n4 = (sh.Nw + 255) / 256;
      const int tiles4 = grid_k4 * grid_n4;
      int z4 = (int)min((long)128, max((long)1, (256 + tiles4 - 1) / (long)tiles4));
      z4 = (int)min((long)z4, max((long)1, sh.L / 64));
      long lpz4 = (sh.L + z4 - 1) / z4;
      lpz4 = ((lpz4 + 63) / 64) * 64;
      z4 = (int)((sh.L + lpz4 - 1) / lpz4);
      if (lpz4 / 64 >= v4_minkt) {
        static bool attr4 = false;
        if (!attr4) {
          (void)hipFuncSetAttribute((const void*)wgrad4_kernel,
                                    hipFuncAttributeMaxDynamicSharedMemorySize,
                                    163840);
          (void)hipFuncSetAttribute((const void*)wgrad5_kernel,
                                    hipFuncAttributeMaxDynamicSharedMemorySize,
                                    163840);
          attr4 = true;
        }
        const size_t lds4 = 8 * (size_t)(128 * 64) * sizeof(bf16);  // 128 KiB
        if (v4 >= 2 || v4 == 3)
          hipLaunchKernelGGL(wgrad5_kernel, dim3(tiles4, z4), dim3(512), lds4,
                             stream, (const bf16*)dy, (const bf16*)x, dw,
                             (const bf16*)zero_page, sh, grid_k4, lpz4);
        else
          hipLaunchKernelGGL(wgrad4_kernel, dim3(tiles4, z4), dim3(512), lds4,
                             stream, (const bf16*)dy, (const bf16*)x, dw,
                             (const bf16*)zero_page, sh, grid_k4, lpz4);
        return;
      }
    }
    const bool narrow_k = K <= 64;
    const int BMK = narrow_k ? 64 : 128, BNW = narrow_k ? 256 : 128;
    const int grid_k = (K + BMK - 1) / BMK;
    const int grid_n = (sh.Nw + BNW - 1) / BNW;
    const int tiles = grid_k * grid_n;
    // split-K: aim for >= 512 blocks to fill 256 CUs (tunable via env)
    static int zcap = -1;
    static long ztarget = -1;
    if (zcap < 0) {
      const char* e = getenv("AL_WGRAD_ZCAP");
      zcap = e ? atoi(e) : 128;
      const char* t = getenv("AL_WGRAD_ZTARGET");
      ztarget = t ? atol(t) : 512;
    }
    long zt = ztarget, zc = zcap;
    if (tiles <= 4) { zt = 2 * ztarget; zc = 4 * zcap; }  // tiny-grid layers
    int z = (int)min(zc, max((long)1, (zt + tiles - 1) / tiles));
    z = (int)min((long)z, max((long)1, sh.L / 64));
    long l_per_z = (sh.L + z - 1) / z;
    l_per_z = ((l_per_z + 63) / 64) * 64;
    z = (int)((sh.L + l_per_z - 1) / l_per_z);
    dim3 grid(tiles, z), block(256);
    static int v2 = -1;
    if (v2 < 0) {
      const char* e = getenv("AL_WGRAD_V2");
      v2 = (e && e[0] == '0') ? 0 : 1;    // tr_b16 staging path (default)
    }
    if (v2) {
      static int scal = -1;
      if (scal < 0) {
        const char* e = getenv("AL_WGRAD_SCAL");
        scal = (e && e[0] == '1') ? 1 : 0;
      }
      static int v3 = -1;
      if (v3 < 0) {
        const char* e = getenv("AL_WGRAD_V3");
        v3 = (e && e[0] == '1') ? 1 : 0;
      }
      if (v3 && !narrow_k) {
        hipLaunchKernelGGL((wgrad3_kernel<2, 2>), grid, dim3(512), 0, stream,
                           (const bf16*)dy, (const bf16*)x, dw,
                           (const bf16*)zero_page, sh, grid_k, l_per_z);
      } else if (scal) {
        if (narrow_k)
          hipLaunchKernelGGL((wgrad2_kernel<1, 4, true>), grid, block, 0, stream,
                             (const bf16*)dy, (const bf16*)x, dw,
                             (const bf16*)zero_page, sh, grid_k, l_per_z);
        else
          hipLaunchKernelGGL((wgrad2_kernel<2, 2, true>), grid, block, 0, stream,
                             (const bf16*)dy, (const bf16*)x, dw,
                             (const bf16*)zero_page, sh, grid_k, l_per_z);
      } else if (narrow_k)
        hipLaunchKernelGGL((wgrad2_kernel<1, 4>), grid, block, 0, stream,
                           (const bf16*)dy, (const bf16*)x, dw,
                           (const bf16*)zero_page, sh, grid_k, l_per_z);
      else
        hipLaunchKernelGGL((wgrad2_kernel<2, 2>), grid, block, 0, stream,
                           (const bf16*)dy, (const bf16*)x, dw,
                           (const bf16*)zero_page, sh, grid_k, l_per_z);
    } else if (narrow_k)
      hipLaunchKernelGGL((wgrad_kernel<1, 4>), grid, block, 0, stream,
                         (const bf16*)dy, (const bf16*)x, dw, sh, grid_k, l_per_z);
    else
      hipLaunchKernelGGL((wgrad_kernel<2, 2>), grid, block, 0, stream,
                         (const bf16*)dy, (const bf16*)x, dw, sh, grid_k, l_per_z);
  } else {
    long total = (long)K * sh.Nw;
    int bx = (int)min((total + 255) / 256, (long)1024);
    int z = (int)min((long)64, max((long)1, sh.L / 8192));
    hipLaunchKernelGGL(wgrad_direct_kernel, dim3(bx, z), dim3(256), 0, stream,
                       (const bf16*)dy, (const bf16*)x, dw, sh);
  }
}
