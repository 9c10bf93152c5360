// Persistent greedy k-center / k-means++ selection kernel.
//
// The greedy loop (reference: coreset_sampler.py:66-105 runs it on the HOST,
// one sync per iteration; round-1 device-side torch loop = 3 kernel launches
// per iteration, ~106 us/iter) becomes ONE cooperative kernel for the whole
// budget: each iteration is a single fused pass (min-update with the last
// selected row + masked argmax / weighted-sample accumulation) bounded by
// grid.sync()s, ~1.5 MB of HBM traffic per iteration at N=130k.
//
// Deterministic mode matches torch argmax tie-breaking (first index wins).
// Randomized (k-means++) mode consumes caller-provided uniforms (one per
// iteration) by inverse-CDF over w_i = max(min_dist_i, 0) (labeled -> 0),
// falling back to uniform-over-unlabeled when the total weight is zero or
// non-finite (reference's NaN-retry jitter, coreset_sampler.py:85-92).
//
// Blocks own contiguous [b*chunk, (b+1)*chunk) ranges so per-block partial
// sums line up with a block-prefix scan done by thread (0,0).

#include "al_common.h"
#include <hip/hip_cooperative_groups.h>

namespace cg = cooperative_groups;

// order-preserving encode: larger score -> larger u64; ties -> smaller index
AL_DEV unsigned long long enc_score(float f, int idx) {
  unsigned int b = __float_as_uint(f);
  b = (b & 0x80000000u) ? ~b : (b | 0x80000000u);
  return ((unsigned long long)b << 32) | (unsigned int)(0x7fffffffu - idx);
}
AL_DEV int dec_idx(unsigned long long v) {
  return 0x7fffffffu - (unsigned int)(v & 0xffffffffu);
}

struct KcState {
  unsigned long long gmax;   // packed (score, idx) argmax cell
  int g_j;                   // last selected index (-1 before first)
  int g_mode;                // 0 = weighted, 1 = uniform fallback
  float g_target;            // remaining CDF mass to consume in pass 2
  int g_found;               // selected index from pass 2
};

template <bool RANDOM>
__global__ void kcenter_kernel(const float* __restrict__ dist,
                               float* __restrict__ min_dist,
                               unsigned char* __restrict__ labeled,
                               long* __restrict__ sel,
                               const float* __restrict__ randu,
                               float* __restrict__ partial,   // [2*gridDim]
                               KcState* __restrict__ st,
                               long n, int iters, int j_init) {
  cg::grid_group grid = cg::this_grid();
  const int nb = gridDim.x;
  const long chunk = (n + nb - 1) / nb;
  const long i0 = (long)blockIdx.x * chunk;
  const long i1 = min(n, i0 + chunk);
  const int tid = threadIdx.x;
  __shared__ float s_red[256 / kWave];
  __shared__ unsigned long long s_max[256 / kWave];
  __shared__ float s_cnt[256 / kWave];

  if (blockIdx.x == 0 && tid == 0) {
    st->gmax = 0ull;
    st->g_j = j_init;
  }
  grid.sync();

  for (int t = 0; t < iters; ++t) {
    const int j = st->g_j;
    const float* row = (j >= 0) ? dist + (long)j * n : nullptr;
    if (RANDOM) {
      // pass 1: min-update + per-block weight sum and unlabeled count
      float wsum = 0.f, cnt = 0.f;
      for (long i = i0 + tid; i < i1; i += blockDim.x) {
        float v = min_dist[i];
        if (row) {
          v = fminf(v, row[i]);
          min_dist[i] = v;
        }
        if (!labeled[i]) {
          wsum += fmaxf(v, 0.f);
          cnt += 1.f;
        }
      }
      wsum = wave_reduce_sum(wsum);
      cnt = wave_reduce_sum(cnt);
      const int wv = tid / kWave, lane = tid % kWave;
      if (lane == 0) { s_red[wv] = wsum; s_cnt[wv] = cnt; }
      __syncthreads();
      if (tid == 0) {
        float a = 0.f, b = 0.f;
        for (int w = 0; w < blockDim.x / kWave; ++w) { a += s_red[w]; b += s_cnt[w]; }
        partial[blockIdx.x] = a;
        partial[nb + blockIdx.x] = b;
      }
      grid.sync();
      // block 0 thread 0: pick mode, locate the target block, leave the
      // in-block residual mass in g_target and the block id in g_found<0 form
      if (blockIdx.x == 0 && tid == 0) {
        float total = 0.f;
        for (int b = 0; b < nb; ++b) total += partial[b];
        int mode = (!__builtin_isfinite(total) || total <= 0.f) ? 1 : 0;
        const float* p = partial + (mode ? nb : 0);
        float tot = total;
        if (mode) {
          tot = 0.f;
          for (int b = 0; b < nb; ++b) tot += p[b];
        }
        float target = randu[t] * tot;
        int tb = nb - 1;
        float acc = 0.f;
        for (int b = 0; b < nb; ++b) {
          if (target < acc + p[b]) { tb = b; target -= acc; break; }
          acc += p[b];
        }
        // target may exceed the tail block's mass by rounding: clamp later
        st->g_mode = mode;
        st->g_target = target;
        st->g_found = -1 - tb;   // negative encoding: block tb scans
      }
      grid.sync();
      // pass 2: the selected block scans its range serially (chunk is tiny)
      if (-1 - st->g_found == blockIdx.x && tid == 0) {
        const int mode = st->g_mode;
        float target = st->g_target;
        long pick = -1, last_ok = -1;
        for (long i = i0; i < i1; ++i) {
          if (labeled[i]) continue;
          const float w = mode ? 1.f : fmaxf(min_dist[i], 0.f);
          if (mode == 0 && w <= 0.f) continue;
          last_ok = i;
          if (target < w) { pick = i; break; }
          target -= w;
        }
        if (pick < 0) pick = last_ok;   // rounding tail: take the last valid
        if (pick < 0) {                 // fully labeled/zero block (degenerate)
          for (long i = 0; i < n; ++i)
            if (!labeled[i]) { pick = i; break; }
        }
        sel[t] = pick;
        labeled[pick] = 1;
        st->g_j = (int)pick;
      }
      grid.sync();
    } else {
      // deterministic: fused min-update + masked argmax
      unsigned long long best = 0ull;
      for (long i = i0 + tid; i < i1; i += blockDim.x) {
        float v = min_dist[i];
        if (row) {
          v = fminf(v, row[i]);
          min_dist[i] = v;
        }
        const float score = labeled[i] ? -3.0e38f : v;
        const unsigned long long e = enc_score(score, (int)i);
        if (e > best) best = e;
      }
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        unsigned long long o =
            (((unsigned long long)(unsigned)__shfl_down((int)(best >> 32), off, 64)) << 32) |
            (unsigned)__shfl_down((int)(best & 0xffffffffu), off, 64);
        if (o > best) best = o;
      }
      const int wv = tid / kWave, lane = tid % kWave;
      if (lane == 0) s_max[wv] = best;
      __syncthreads();
      if (tid == 0) {
        for (int w = 1; w < blockDim.x / kWave; ++w)
          if (s_max[w] > best) best = s_max[w];
        atomicMax((unsigned long long*)&st->gmax, best);
      }
      grid.sync();
      if (blockIdx.x == 0 && tid == 0) {
        const int jj = dec_idx(st->gmax);
        sel[t] = jj;
        labeled[jj] = 1;
        st->g_j = jj;
        st->gmax = 0ull;
      }
      grid.sync();
    }
  }
}

extern "C" int al_kcenter_greedy(const float* dist, float* min_dist,
                                 unsigned char* labeled, long* sel,
                                 const float* randu, float* partial, void* st,
                                 long n, int iters, int j_init, int randomize,
                                 int nblocks, hipStream_t stream) {
  int dev = 0, coop = 0;
  (void)hipGetDevice(&dev);
  (void)hipDeviceGetAttribute(&coop, hipDeviceAttributeCooperativeLaunch, dev);
  if (!coop) return -1;
  dim3 grid(nblocks), block(256);
  void* args[] = {(void*)&dist, (void*)&min_dist, (void*)&labeled, (void*)&sel,
                  (void*)&randu, (void*)&partial, (void*)&st, (void*)&n,
                  (void*)&iters, (void*)&j_init};
  hipError_t err;
  if (randomize)
    err = hipLaunchCooperativeKernel((const void*)kcenter_kernel<true>, grid,
                                     block, args, 0, stream);
  else
    err = hipLaunchCooperativeKernel((const void*)kcenter_kernel<false>, grid,
                                     block, args, 0, stream);
  return err == hipSuccess ? 0 : -2;
}
