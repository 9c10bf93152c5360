// Fused query-scoring + cross-entropy kernels over logits (B, C) fp32.
//
// softmax_scores: one pass -> (top1 prob, margin = p1 - p2, entropy) per row
//   (reference does softmax + topk as separate ATen ops,
//    confidence_sampler.py:31-33 / margin_sampler.py:33-35).
// ce_fwd / ce_bwd: fused softmax + NLL (+ class-weighted scaling handled by
//   the Python wrapper), reference nn.CrossEntropyLoss (strategy.py:352-356).
// One wave per row; C up to a few thousand, strided by 64 lanes.

#include "al_common.h"

__global__ void softmax_scores_kernel(const float* __restrict__ logits,
                                      float* __restrict__ out, int B, int C) {
  const int row = blockIdx.x * (blockDim.x / kWave) + (threadIdx.x / kWave);
  if (row >= B) return;
  const int lane = threadIdx.x % kWave;
  const float* l = logits + (long)row * C;

  // per-lane top-2 logits
  float m1 = -3.0e38f, m2 = -3.0e38f;
  for (int c = lane; c < C; c += kWave) {
    const float v = l[c];
    if (v > m1) { m2 = m1; m1 = v; }
    else if (v > m2) m2 = v;
  }
  // wave-combine top-2
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float o1 = __shfl_down(m1, off, 64), o2 = __shfl_down(m2, off, 64);
    if (o1 > m1) { m2 = fmaxf(m1, o2); m1 = o1; }
    else m2 = fmaxf(m2, o1);
  }
  m1 = wave_bcast(m1, 0);
  m2 = wave_bcast(m2, 0);

  // denominator + entropy accumulation:
  //   Z = sum exp(v - m1);  H = log Z - (1/Z) * sum (v - m1) exp(v - m1)
  float z = 0.f, se = 0.f;
  for (int c = lane; c < C; c += kWave) {
    const float d = l[c] - m1;
    const float e = __expf(d);
    z += e;
    se += d * e;
  }
  z = wave_bcast(wave_reduce_sum(z), 0);
  se = wave_bcast(wave_reduce_sum(se), 0);
  if (lane == 0) {
    const float p1 = 1.0f / z;                    // exp(m1-m1)/Z
    const float p2 = __expf(m2 - m1) / z;
    const float ent = __logf(z) - se / z;
    out[row] = p1;
    out[B + row] = p1 - p2;
    out[2 * B + row] = ent;
  }
}

extern "C" void al_softmax_scores(const float* logits, float* out, int B, int C,
                                  hipStream_t stream) {
  const int waves_per_block = 4;
  dim3 block(kWave * waves_per_block);
  dim3 grid((B + waves_per_block - 1) / waves_per_block);
  hipLaunchKernelGGL(softmax_scores_kernel, grid, block, 0, stream, logits, out, B, C);
}

// ---------------------------------------------------------------------------

__global__ void ce_fwd_kernel(const float* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ losses, float* __restrict__ probs,
                              int B, int C) {
  const int row = blockIdx.x * (blockDim.x / kWave) + (threadIdx.x / kWave);
  if (row >= B) return;
  const int lane = threadIdx.x % kWave;
  const float* l = logits + (long)row * C;
  float m = -3.0e38f;
  for (int c = lane; c < C; c += kWave) m = fmaxf(m, l[c]);
  m = wave_bcast(wave_reduce_max(m), 0);
  float z = 0.f;
  for (int c = lane; c < C; c += kWave) z += __expf(l[c] - m);
  z = wave_bcast(wave_reduce_sum(z), 0);
  const float invz = 1.0f / z;
  float* p = probs + (long)row * C;
  for (int c = lane; c < C; c += kWave) p[c] = __expf(l[c] - m) * invz;
  if (lane == 0) {
    const long t = targets[row];
    losses[row] = __logf(z) - (l[t] - m);
  }
}

extern "C" void al_ce_fwd(const float* logits, const long* targets, float* losses,
                          float* probs, int B, int C, hipStream_t stream) {
  const int wpb = 4;
  hipLaunchKernelGGL(ce_fwd_kernel, dim3((B + wpb - 1) / wpb), dim3(kWave * wpb), 0,
                     stream, logits, targets, losses, probs, B, C);
}

__global__ void ce_bwd_kernel(const float* __restrict__ probs,
                              const long* __restrict__ targets,
                              const float* __restrict__ scale,
                              float* __restrict__ dlogits, int B, int C) {
  const long total = (long)B * C;
  for (long i = grid_stride_begin(); i < total; i += grid_stride_step()) {
    const int row = (int)(i / C);
    const int c = (int)(i % C);
    const float s = scale[row];
    float v = probs[i] * s;
    if (c == (int)targets[row]) v -= s;
    dlogits[i] = v;
  }
}

extern "C" void al_ce_bwd(const float* probs, const long* targets, const float* scale,
                          float* dlogits, int B, int C, hipStream_t stream) {
  long total = (long)B * C;
  int blocks = (int)min((total + 255) / 256, (long)2048);
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(blocks), dim3(256), 0, stream, probs, targets,
                     scale, dlogits, B, C);
}
