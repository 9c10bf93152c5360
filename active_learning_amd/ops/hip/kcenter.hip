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
      // EVERY block's wave 0 redundantly computes the CDF prefix over the
      // <=512 per-block partials and decides whether the target falls in
      // ITS range — identical arithmetic on identical inputs picks exactly
      // one owner with NO second grid.sync (the former block-0-then-sync
      // scheme cost one more sync per iteration, ~30% of the loop).
      __shared__ int s_tb;
      __shared__ float s_target;
      __shared__ int s_mode;
      if (tid == 0) s_tb = -1;
      __syncthreads();
      if (tid < kWave) {
        const int lane = tid;
        float wsum = 0.f, csum = 0.f;
        for (int b = lane; b < nb; b += kWave) {
          wsum += partial[b];
          csum += partial[nb + b];
        }
        float total = wave_bcast(wave_reduce_sum(wsum), 0);
        const float ctotal = wave_bcast(wave_reduce_sum(csum), 0);
        const int mode = (!__builtin_isfinite(total) || total <= 0.f) ? 1 : 0;
        const float* p = partial + (mode ? nb : 0);
        const float tot = mode ? ctotal : total;
        float target = randu[t] * tot;
        // sequential over ceil(nb/64) strips, wave-scan within each
        int tb = -1;
        float acc = 0.f;
        for (int s0 = 0; s0 < nb && tb < 0; s0 += kWave) {
          const int b = s0 + lane;
          float v = (b < nb) ? p[b] : 0.f;
          // inclusive wave prefix sum
          float pref = v;
#pragma unroll
          for (int off = 1; off < kWave; off <<= 1) {
            const float o = __shfl_up(pref, off, 64);
            if (lane >= off) pref += o;
          }
          const float strip = wave_bcast(pref, kWave - 1);
          // first lane whose [acc+pref-v, acc+pref) contains target
          const bool hit = (b < nb) && (target < acc + pref) &&
                           (target >= acc + pref - v);
          const unsigned long long ball = __ballot(hit);
          if (ball) {
            const int l = __ffsll((unsigned long long)ball) - 1;
            const float pref_l = wave_bcast(pref, l);
            const float v_l = wave_bcast(v, l);
            tb = s0 + l;
            target -= acc + pref_l - v_l;
          }
          acc += strip;
        }
        if (tb < 0) tb = nb - 1;  // rounding tail
        if (lane == 0) {
          s_tb = tb;
          s_target = target;
          s_mode = mode;
        }
      }
      __syncthreads();
      // pass 2: the owning block finds its in-range index with a
      // block-parallel prefix (per-thread contiguous sub-ranges, LDS scan)
      if (s_tb == (int)blockIdx.x) {
        if (tid == 0) st->g_found = 0;  // owner-only: reset claim flag
        __syncthreads();
        const int mode = s_mode;
        const long span = i1 - i0;
        const long per = (span + blockDim.x - 1) / blockDim.x;
        const long j0 = i0 + tid * per;
        const long j1 = min(i1, j0 + per);
        float mysum = 0.f;
        for (long i = j0; i < j1; ++i) {
          if (labeled[i]) continue;
          mysum += mode ? 1.f : fmaxf(min_dist[i], 0.f);
        }
        __shared__ float s_pref[256];
        s_pref[tid] = mysum;
        __syncthreads();
        // simple Hillis-Steele inclusive scan over 256 entries
        for (int off = 1; off < (int)blockDim.x; off <<= 1) {
          float add = (tid >= off) ? s_pref[tid - off] : 0.f;
          __syncthreads();
          s_pref[tid] += add;
          __syncthreads();
        }
        const float target = s_target;
        const float excl = (tid == 0) ? 0.f : s_pref[tid - 1];
        const bool mine = (mysum > 0.f) && (target >= excl) &&
                          (target < excl + mysum);
        const bool tail = (tid == (int)blockDim.x - 1) &&
                          (target >= s_pref[blockDim.x - 1]);
        if (mine || (tail && mysum > 0.f)) {
          float rem = mine ? (target - excl) : mysum * 0.999f;
          long pick = -1, last_ok = -1;
          for (long i = j0; i < j1; ++i) {
            if (labeled[i]) continue;
            const float w = mode ? 1.f : fmaxf(min_dist[i], 0.f);
            if (mode == 0 && w <= 0.f) continue;
            last_ok = i;
            if (rem < w) { pick = i; break; }
            rem -= w;
          }
          if (pick < 0) pick = last_ok;
          if (pick >= 0) {
            sel[t] = pick;
            labeled[pick] = 1;
            st->g_j = (int)pick;
            st->g_found = 1;  // claimed (owner-block scope)
          }
        }
        __syncthreads();
        if (tid == 0 && st->g_found != 1) {
          // degenerate: every sub-range empty — fall back to first unlabeled
          for (long i = 0; i < n; ++i)
            if (!labeled[i]) {
              sel[t] = i;
              labeled[i] = 1;
              st->g_j = (int)i;
              break;
            }
        }
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
