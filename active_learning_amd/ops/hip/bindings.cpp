// Torch binding layer for the gfx950 HIP kernels (active_learning_amd._C).
//
// This TU is compiled by the host C++ compiler against the PyTorch-ROCm
// headers; all device code lives in the *.hip TUs (compiled directly by
// hipcc --offload-arch=gfx950 — no hipify anywhere) and is reached through
// the extern "C" launchers declared below.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <algorithm>
#include <vector>

using torch::Tensor;

extern "C" {
void al_conv2d_mm(int mode, const void* A, const void* B, void* out,
                  const void* zero_page, int N, int H, int W, int C, int K, int R,
                  int S, int P, int Q, int stride, int pad, const float* epi_scale,
                  const float* epi_shift, const void* epi_res, int epi_relu,
                  float* stat_sum, float* stat_sumsq, const void* bnb_mask,
                  const void* bnb_x, const float* bnb_mean,
                  const float* bnb_invstd, void* stream);
void al_conv2d_wgrad(const void* dy, const void* x, float* dw,
                     const void* zero_page, int N, int H, int W,
                     int C, int K, int R, int S, int P, int Q, int stride, int pad,
                     void* stream);
void al_bn_stats(const void* x, float* sum, float* sumsq, long rows, int C,
                 void* stream);
int al_bn_reduce_blocks(long rows, int C);
void al_bn_part_reduce(const float* part_a, const float* part_b, float* out_a,
                       float* out_b, int nb, int C, void* stream);
void al_bn_finalize(const float* part_s, const float* part_ss, float* mean,
                    float* invstd, float* running_mean, float* running_var, int nb,
                    int C, float n, float momentum, float eps, int update_running,
                    void* stream);
void al_bn_norm_fwd(const void* x, void* y, const float* mean, const float* invstd,
                    const float* gamma, const float* beta, const void* res, int relu,
                    void* relu_mask, long rows, int C, void* stream);
void al_bn_bwd_reduce(const void* dy, const void* x, const void* relu_mask,
                      const float* mean,
                      const float* invstd, float* sum_dy, float* sum_dy_xhat, int relu,
                      long rows, int C, void* stream);
void al_bn_bwd(const void* dy, const void* x, const void* relu_mask, const float* mean,
               const float* invstd, const float* gamma, const float* sum_dy,
               const float* sum_dy_xhat, float n, int use_batch_stats, int relu,
               int has_res, void* dx, void* dres, long rows, int C, void* stream);
void al_maxpool_fwd(const void* x, void* y, int* idx, int N, int H, int W, int C,
                    int P, int Q, int kernel, int stride, int pad, void* stream);
void al_maxpool_bwd(const void* dy, const int* idx, void* dx, int N, int H, int W,
                    int C, int P, int Q, int kernel, int stride, int pad, void* stream);
void al_global_avg_pool(const void* x, void* y, int N, int HW, int C, void* stream);
void al_softmax_scores(const float* logits, float* out, int B, int C, void* stream);
void al_ce_fwd(const float* logits, const long* targets, float* losses, float* probs,
               int B, int C, void* stream);
void al_ce_bwd(const float* probs, const long* targets, const float* scale,
               float* dlogits, int B, int C, void* stream);
void al_im2col_pack(const void* x, void* out, int N, int H, int W, int C, int R,
                    int S, int P, int Q, int stride, int pad, int kdpad,
                    int rowpad, void* stream);
void al_sgd_step(float* p, const float* g, float* buf, float lr, float momentum,
                 float wd, long n, void* shadow, void* stream);
void al_adam_step(float* p, const float* g, float* m, float* v, float lr, float b1,
                  float b2, float eps, float wd, float bc1, float bc2, long n,
                  void* stream);
void al_sgd_step_multi(const void* table, int nchunks, float lr, float momentum,
                       float wd, int zero_grad, void* stream);
void al_wt_refresh(const void* table, int nrows, void* stream);
void al_sgd_step_multi_dev(const void* table, int nchunks, const float* hyper,
                           int zero_grad, void* stream);
int al_kcenter_greedy(const float* dist, float* min_dist, unsigned char* labeled,
                      long* sel, const float* randu, float* partial, void* st,
                      long n, int iters, int j_init, int randomize, int nblocks,
                      void* stream);
void al_scatter_s2(const void* tmp, void* dx, int N, int H, int W, int C, int P,
                   int Q, int stride, void* stream);
void al_badge_gram(const float* a, const float* e, const float* d, float* out,
                   long N, int Ka, int Ke, void* stream);
int al_pairwise_sqdist(const void* f, const float* sq, float* out,
                       const void* zero, long N, int M, void* stream);
int al_linear_needs_zero(int M_, int N_, int K_);
void al_linear_fwd(const float* x, const float* w, const float* bias, float* out,
                   int B, int M, int C, void* stream);
void al_linear_bwd(const float* dy, const float* x, const float* w, float* dx,
                   float* dw, float* db, int B, int M, int C, void* stream);
}

namespace {

void* cur_stream() { return (void*)c10::hip::getCurrentHIPStream().stream(); }

void check_bf16_contig(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name,
              " must be bf16 on the GPU kernel path");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

const Tensor& zero_page(const Tensor& like) {
  static Tensor zp;
  if (!zp.defined())
    zp = torch::zeros({4096}, like.options().dtype(torch::kBFloat16));
  return zp;
}

int out_dim(int in, int k, int stride, int pad) {
  return (in + 2 * pad - k) / stride + 1;
}

Tensor conv2d_fwd(const Tensor& x, const Tensor& w, long stride, long pad) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int K = w.size(0), R = w.size(1), S = w.size(2);
  TORCH_CHECK(w.size(3) == C, "conv2d_fwd: channel mismatch");
  const int P = out_dim(H, R, stride, pad), Q = out_dim(W, S, stride, pad);
  auto y = torch::empty({N, P, Q, K}, x.options());
  al_conv2d_mm(0, x.data_ptr(), w.data_ptr(), y.data_ptr(),
               zero_page(x).data_ptr(), N, H, W, C, K, R, S, P, Q, (int)stride,
               (int)pad, nullptr, nullptr, nullptr, 0, nullptr, nullptr, nullptr, nullptr, nullptr, nullptr,
               cur_stream());
  return y;
}

std::vector<Tensor> conv2d_fwd_stats(const Tensor& x, const Tensor& w, long stride,
                                     long pad, const Tensor& both_in) {
  // training-path fusion: conv + per-column sum/sumsq of the output in the
  // same kernel (feeds the following BatchNorm's batch statistics).
  // both_in: optional PRE-ZEROED 2K fp32 slot from the python stats arena
  // (one flat zero fill per step instead of one per conv); empty -> allocate.
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int K = w.size(0), R = w.size(1), S = w.size(2);
  TORCH_CHECK(w.size(3) == C, "conv2d_fwd_stats: channel mismatch");
  const int P = out_dim(H, R, stride, pad), Q = out_dim(W, S, stride, pad);
  auto y = torch::empty({N, P, Q, K}, x.options());
  auto opts = x.options().dtype(torch::kFloat32);
  Tensor both = both_in.numel() ? both_in : torch::zeros({2L * K}, opts);
  TORCH_CHECK(both.numel() >= 2L * K && both.scalar_type() == torch::kFloat32);
  auto sum = both.narrow(0, 0, K);
  auto sumsq = both.narrow(0, K, K);
  al_conv2d_mm(0, x.data_ptr(), w.data_ptr(), y.data_ptr(),
               zero_page(x).data_ptr(), N, H, W, C, K, R, S, P, Q, (int)stride,
               (int)pad, nullptr, nullptr, nullptr, 0, sum.data_ptr<float>(),
               sumsq.data_ptr<float>(), nullptr, nullptr, nullptr, nullptr, cur_stream());
  return {y, sum, sumsq};
}

Tensor conv2d_fwd_fused(const Tensor& x, const Tensor& w, long stride, long pad,
                        const Tensor& scale, const Tensor& shift, bool relu,
                        const Tensor& residual) {
  // inference path: y = [relu](conv(x,w) * scale + shift [+ residual]) —
  // frozen-stats BN folded into the conv epilogue (query/eval passes)
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int K = w.size(0), R = w.size(1), S = w.size(2);
  TORCH_CHECK(w.size(3) == C, "conv2d_fwd_fused: channel mismatch");
  TORCH_CHECK(scale.numel() == K && shift.numel() == K);
  const int P = out_dim(H, R, stride, pad), Q = out_dim(W, S, stride, pad);
  auto y = torch::empty({N, P, Q, K}, x.options());
  const bool has_res = residual.numel() > 0;
  if (has_res) check_bf16_contig(residual, "residual");
  al_conv2d_mm(0, x.data_ptr(), w.data_ptr(), y.data_ptr(),
               zero_page(x).data_ptr(), N, H, W, C, K, R, S, P, Q, (int)stride,
               (int)pad, scale.contiguous().data_ptr<float>(),
               shift.contiguous().data_ptr<float>(),
               has_res ? residual.data_ptr() : nullptr, relu ? 1 : 0, nullptr,
               nullptr, nullptr, nullptr, nullptr, nullptr, cur_stream());
  return y;
}

Tensor conv2d_bwd_data(const Tensor& dy, const Tensor& wt, long stride, long pad,
                       long H, long W) {
  // wt is the weight PRE-PERMUTED to (C,R,S,K) — k fastest in the contraction;
  // the Python layer caches this permutation per weight version.
  check_bf16_contig(dy, "dy");
  check_bf16_contig(wt, "wt");
  const int N = dy.size(0), P = dy.size(1), Q = dy.size(2), K = dy.size(3);
  const int C = wt.size(0), R = wt.size(1), S = wt.size(2);
  TORCH_CHECK(wt.size(3) == K, "conv2d_bwd_data: channel mismatch (wt must be CRSK)");
  auto dx = torch::empty({N, (long)H, (long)W, C}, dy.options());
  al_conv2d_mm(1, dy.data_ptr(), wt.data_ptr(), dx.data_ptr(),
               zero_page(dy).data_ptr(), N, (int)H, (int)W, C, K, R, S, P, Q,
               (int)stride, (int)pad, nullptr, nullptr, nullptr, 0, nullptr, nullptr, nullptr, nullptr, nullptr, nullptr,
               cur_stream());
  return dx;
}

Tensor conv2d_bwd_data_res(const Tensor& dy, const Tensor& wt, long stride,
                           long pad, long H, long W, const Tensor& res) {
  // bwd-data with the upstream residual GRADIENT accumulated in the
  // epilogue: dx = conv_bwd(dy) + res — replaces autograd's fan-in add pass
  // for residual joins (ops/functional.py RESBACK side channel).
  check_bf16_contig(dy, "dy");
  check_bf16_contig(wt, "wt");
  check_bf16_contig(res, "res");
  const int N = dy.size(0), P = dy.size(1), Q = dy.size(2), K = dy.size(3);
  const int C = wt.size(0), R = wt.size(1), S = wt.size(2);
  TORCH_CHECK(wt.size(3) == K, "channel mismatch (wt must be CRSK)");
  TORCH_CHECK(res.numel() == (long)N * H * W * C, "res size");
  auto dx = torch::empty({N, (long)H, (long)W, C}, dy.options());
  al_conv2d_mm(1, dy.data_ptr(), wt.data_ptr(), dx.data_ptr(),
               zero_page(dy).data_ptr(), N, (int)H, (int)W, C, K, R, S, P, Q,
               (int)stride, (int)pad, nullptr, nullptr, res.data_ptr(), 0,
               nullptr, nullptr, nullptr, nullptr, nullptr, nullptr,
               cur_stream());
  return dx;
}

std::vector<Tensor> conv2d_bwd_data_bn(const Tensor& dy, const Tensor& wt,
                                       long stride, long pad, long H, long W,
                                       const Tensor& mask, const Tensor& xbn,
                                       const Tensor& mean, const Tensor& invstd) {
  // bwd-data with the upstream BatchNorm's backward reduction fused into the
  // epilogue: dx is returned PRE-MASKED (dy~ = relu_mask * dx) along with
  // per-channel (sum dy~, sum dy~*xhat) — bn_bwd_reduce never runs.
  check_bf16_contig(dy, "dy");
  check_bf16_contig(wt, "wt");
  check_bf16_contig(xbn, "xbn");
  const int N = dy.size(0), P = dy.size(1), Q = dy.size(2), K = dy.size(3);
  const int C = wt.size(0), R = wt.size(1), S = wt.size(2);
  TORCH_CHECK(wt.size(3) == K, "channel mismatch (wt must be CRSK)");
  TORCH_CHECK(mask.scalar_type() == torch::kUInt8 && mask.is_contiguous());
  TORCH_CHECK(mask.numel() * 8 == (long)N * H * W * C, "mask size");
  TORCH_CHECK(xbn.numel() == (long)N * H * W * C, "xbn size");
  TORCH_CHECK(mean.scalar_type() == torch::kFloat32 && mean.numel() == C);
  TORCH_CHECK(invstd.scalar_type() == torch::kFloat32 && invstd.numel() == C);
  auto dx = torch::empty({N, (long)H, (long)W, C}, dy.options());
  auto both = torch::zeros({2L * C}, dy.options().dtype(torch::kFloat32));
  auto sum = both.narrow(0, 0, C);
  auto sumx = both.narrow(0, C, C);
  al_conv2d_mm(1, dy.data_ptr(), wt.data_ptr(), dx.data_ptr(),
               zero_page(dy).data_ptr(), N, (int)H, (int)W, C, K, R, S, P, Q,
               (int)stride, (int)pad, nullptr, nullptr, nullptr, 0,
               sum.data_ptr<float>(), sumx.data_ptr<float>(), mask.data_ptr(),
               xbn.data_ptr(), mean.data_ptr<float>(),
               invstd.data_ptr<float>(), cur_stream());
  return {dx, sum, sumx};
}

Tensor conv2d_bwd_weight(const Tensor& dy, const Tensor& x, long R, long S,
                         long stride, long pad) {
  check_bf16_contig(dy, "dy");
  check_bf16_contig(x, "x");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int P = dy.size(1), Q = dy.size(2), K = dy.size(3);
  auto dw = torch::zeros({(long)K, R, S, (long)C},
                         x.options().dtype(torch::kFloat32));
  al_conv2d_wgrad(dy.data_ptr(), x.data_ptr(), dw.data_ptr<float>(),
                  zero_page(x).data_ptr(), N, H, W, C, K,
                  (int)R, (int)S, P, Q, (int)stride, (int)pad, cur_stream());
  return dw;
}

Tensor conv2d_bwd_weight_into(const Tensor& dy, const Tensor& x, long R, long S,
                              long stride, long pad, Tensor out) {
  // accumulate into a caller-provided (pre-zeroed) fp32 buffer: lets the
  // grad arena replace ~50 per-conv zero fills per step with one
  check_bf16_contig(dy, "dy");
  check_bf16_contig(x, "x");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int P = dy.size(1), Q = dy.size(2), K = dy.size(3);
  TORCH_CHECK(out.is_cuda() && out.is_contiguous() &&
              out.scalar_type() == torch::kFloat32 &&
              out.numel() == (long)K * R * S * C,
              "conv2d_bwd_weight_into: bad out buffer");
  al_conv2d_wgrad(dy.data_ptr(), x.data_ptr(), out.data_ptr<float>(),
                  zero_page(x).data_ptr(), N, H, W, C, K,
                  (int)R, (int)S, P, Q, (int)stride, (int)pad, cur_stream());
  return out;
}

Tensor im2col_pack(const Tensor& x, long R, long S, long stride, long pad,
                   long kdpad, long rowpad) {
  check_bf16_contig(x, "x");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int P = out_dim(H, R, stride, pad), Q = out_dim(W, S, stride, pad);
  TORCH_CHECK(rowpad % 8 == 0 && rowpad >= S * C && R * rowpad <= kdpad);
  auto out = torch::empty({N, P, Q, kdpad}, x.options());
  al_im2col_pack(x.data_ptr(), out.data_ptr(), N, H, W, C, (int)R, (int)S, P, Q,
                 (int)stride, (int)pad, (int)kdpad, (int)rowpad, cur_stream());
  return out;
}

std::vector<Tensor> bn_stats_finalize(const Tensor& x, Tensor& running_mean,
                                      Tensor& running_var, double momentum,
                                      double eps, bool update_running) {
  // single-process fast path: partial sums -> mean/invstd (+ running update)
  // without any host-side scalar math or small ATen kernels
  check_bf16_contig(x, "x");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  auto opts = x.options().dtype(torch::kFloat32);
  const int nb = al_bn_reduce_blocks(rows, C);
  auto part_sum = torch::empty({nb, C}, opts);
  auto part_sumsq = torch::empty({nb, C}, opts);
  al_bn_stats(x.data_ptr(), part_sum.data_ptr<float>(), part_sumsq.data_ptr<float>(),
              rows, C, cur_stream());
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  al_bn_finalize(part_sum.data_ptr<float>(), part_sumsq.data_ptr<float>(),
                 mean.data_ptr<float>(), invstd.data_ptr<float>(),
                 running_mean.data_ptr<float>(), running_var.data_ptr<float>(), nb, C,
                 (float)rows, (float)momentum, (float)eps, update_running ? 1 : 0,
                 cur_stream());
  return {mean, invstd};
}

std::vector<Tensor> bn_finalize(const Tensor& s, const Tensor& ss, Tensor& rm,
                                Tensor& rv, double n, double momentum, double eps,
                                bool update_running) {
  const int C = s.numel();
  auto opts = s.options();
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  al_bn_finalize(s.contiguous().data_ptr<float>(), ss.contiguous().data_ptr<float>(),
                 mean.data_ptr<float>(), invstd.data_ptr<float>(),
                 rm.data_ptr<float>(), rv.data_ptr<float>(), 1, C, (float)n,
                 (float)momentum, (float)eps, update_running ? 1 : 0, cur_stream());
  return {mean, invstd};
}

std::vector<Tensor> bn_stats(const Tensor& x) {
  check_bf16_contig(x, "x");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  auto opts = x.options().dtype(torch::kFloat32);
  const int nb = al_bn_reduce_blocks(rows, C);
  auto part_sum = torch::empty({nb, C}, opts);
  auto part_sumsq = torch::empty({nb, C}, opts);
  al_bn_stats(x.data_ptr(), part_sum.data_ptr<float>(), part_sumsq.data_ptr<float>(),
              rows, C, cur_stream());
  auto sum = torch::empty({C}, opts);
  auto sumsq = torch::empty({C}, opts);
  al_bn_part_reduce(part_sum.data_ptr<float>(), part_sumsq.data_ptr<float>(),
                    sum.data_ptr<float>(), sumsq.data_ptr<float>(), nb, C,
                    cur_stream());
  return {sum, sumsq};
}

std::vector<Tensor> bn_norm_fwd(const Tensor& x, const Tensor& mean,
                                const Tensor& invstd,
                                const Tensor& gamma, const Tensor& beta, bool relu,
                                const Tensor& residual, bool want_mask) {
  check_bf16_contig(x, "x");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  TORCH_CHECK(C % 8 == 0, "bn_norm_fwd: C % 8 required");
  auto y = torch::empty_like(x);
  const bool has_res = residual.numel() > 0;
  if (has_res) check_bf16_contig(residual, "residual");
  // relu mask: one bit per element, read by the backward passes in place of
  // a full re-stream of y
  Tensor mask = (relu && want_mask)
                    ? torch::empty({rows, (long)C / 8},
                                   x.options().dtype(torch::kUInt8))
                    : torch::empty({0}, x.options().dtype(torch::kUInt8));
  al_bn_norm_fwd(x.data_ptr(), y.data_ptr(), mean.contiguous().data_ptr<float>(),
                 invstd.contiguous().data_ptr<float>(),
                 gamma.contiguous().data_ptr<float>(),
                 beta.contiguous().data_ptr<float>(),
                 has_res ? residual.data_ptr() : nullptr, relu ? 1 : 0,
                 mask.numel() ? mask.data_ptr() : nullptr, rows, C,
                 cur_stream());
  return {y, mask};
}

std::vector<Tensor> bn_bwd_reduce(const Tensor& dy, const Tensor& x,
                                  const Tensor& relu_mask,
                                  const Tensor& mean, const Tensor& invstd,
                                  bool relu) {
  TORCH_CHECK(!relu || relu_mask.numel() > 0, "bn_bwd_reduce: relu needs mask");
  check_bf16_contig(dy, "dy");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  auto opts = x.options().dtype(torch::kFloat32);
  const int nb = al_bn_reduce_blocks(rows, C);
  auto part_s = torch::empty({nb, C}, opts);
  auto part_sx = torch::empty({nb, C}, opts);
  al_bn_bwd_reduce(dy.data_ptr(), x.data_ptr(),
                   relu_mask.numel() ? relu_mask.data_ptr() : nullptr,
                   mean.contiguous().data_ptr<float>(),
                   invstd.contiguous().data_ptr<float>(), part_s.data_ptr<float>(),
                   part_sx.data_ptr<float>(), relu ? 1 : 0, rows, C,
                   cur_stream());
  auto sum_dy = torch::empty({C}, opts);
  auto sum_dy_xhat = torch::empty({C}, opts);
  al_bn_part_reduce(part_s.data_ptr<float>(), part_sx.data_ptr<float>(),
                    sum_dy.data_ptr<float>(), sum_dy_xhat.data_ptr<float>(), nb, C,
                    cur_stream());
  return {sum_dy, sum_dy_xhat};
}

std::vector<Tensor> bn_bwd(const Tensor& dy, const Tensor& x,
                           const Tensor& relu_mask,
                           const Tensor& mean, const Tensor& invstd,
                           const Tensor& gamma, const Tensor& sum_dy,
                           const Tensor& sum_dy_xhat, double n, bool use_batch_stats,
                           bool relu, bool has_res) {
  check_bf16_contig(dy, "dy");
  TORCH_CHECK(!relu || relu_mask.numel() > 0, "bn_bwd: relu needs mask");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  auto dx = torch::empty_like(x);
  Tensor dres;
  if (has_res) dres = torch::empty_like(x);
  al_bn_bwd(dy.data_ptr(), x.data_ptr(),
            relu_mask.numel() ? relu_mask.data_ptr() : nullptr,
            mean.contiguous().data_ptr<float>(),
            invstd.contiguous().data_ptr<float>(),
            gamma.contiguous().data_ptr<float>(),
            sum_dy.contiguous().data_ptr<float>(),
            sum_dy_xhat.contiguous().data_ptr<float>(), (float)n,
            use_batch_stats ? 1 : 0, relu ? 1 : 0, has_res ? 1 : 0, dx.data_ptr(),
            has_res ? dres.data_ptr() : nullptr, rows, C, cur_stream());
  if (!has_res) dres = torch::empty({0}, x.options());
  return {dx, dres};
}

std::vector<Tensor> maxpool2d_fwd(const Tensor& x, long kernel, long stride,
                                  long pad) {
  check_bf16_contig(x, "x");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int P = out_dim(H, kernel, stride, pad), Q = out_dim(W, kernel, stride, pad);
  auto y = torch::empty({N, P, Q, C}, x.options());
  auto idx = torch::empty({N, P, Q, C}, x.options().dtype(torch::kInt32));
  al_maxpool_fwd(x.data_ptr(), y.data_ptr(), idx.data_ptr<int>(), N, H, W, C, P, Q,
                 (int)kernel, (int)stride, (int)pad, cur_stream());
  return {y, idx};
}

Tensor maxpool2d_bwd(const Tensor& dy, const Tensor& idx, long H, long W, long kernel,
                     long stride, long pad) {
  check_bf16_contig(dy, "dy");
  const int N = dy.size(0), P = dy.size(1), Q = dy.size(2), C = dy.size(3);
  auto dx = torch::empty({N, (long)H, (long)W, C}, dy.options());
  al_maxpool_bwd(dy.data_ptr(), idx.data_ptr<int>(), dx.data_ptr(), N, (int)H,
                 (int)W, C, P, Q, (int)kernel, (int)stride, (int)pad, cur_stream());
  return dx;
}

Tensor global_avg_pool(const Tensor& x) {
  check_bf16_contig(x, "x");
  const int N = x.size(0), HW = x.size(1) * x.size(2), C = x.size(3);
  auto y = torch::empty({N, C}, x.options());
  al_global_avg_pool(x.data_ptr(), y.data_ptr(), N, HW, C, cur_stream());
  return y;
}

Tensor softmax_scores(const Tensor& logits) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2);
  auto l = logits.to(torch::kFloat32).contiguous();
  const int B = l.size(0), C = l.size(1);
  auto out = torch::empty({3, B}, l.options());
  al_softmax_scores(l.data_ptr<float>(), out.data_ptr<float>(), B, C, cur_stream());
  return out;
}

std::vector<Tensor> ce_fwd(const Tensor& logits, const Tensor& targets,
                           const Tensor& /*weights: handled in python*/) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2);
  auto l = logits.contiguous();
  TORCH_CHECK(l.scalar_type() == torch::kFloat32, "ce_fwd expects fp32 logits");
  const int B = l.size(0), C = l.size(1);
  auto t = targets.to(torch::kInt64).contiguous();
  auto losses = torch::empty({B}, l.options());
  auto probs = torch::empty({B, C}, l.options());
  al_ce_fwd(l.data_ptr<float>(), t.data_ptr<long>(), losses.data_ptr<float>(),
            probs.data_ptr<float>(), B, C, cur_stream());
  return {losses, probs};
}

Tensor ce_bwd(const Tensor& probs, const Tensor& targets, const Tensor& scale) {
  const int B = probs.size(0), C = probs.size(1);
  auto dl = torch::empty_like(probs);
  al_ce_bwd(probs.contiguous().data_ptr<float>(),
            targets.to(torch::kInt64).contiguous().data_ptr<long>(),
            scale.contiguous().data_ptr<float>(), dl.data_ptr<float>(), B, C,
            cur_stream());
  return dl;
}

void sgd_step(Tensor& p, const Tensor& g, Tensor& buf, double lr, double momentum,
              double wd, const Tensor& shadow) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kFloat32);
  void* sh = nullptr;
  if (shadow.numel()) {
    TORCH_CHECK(shadow.numel() == p.numel() &&
                shadow.scalar_type() == torch::kBFloat16);
    sh = shadow.data_ptr();
  }
  al_sgd_step(p.data_ptr<float>(), g.contiguous().data_ptr<float>(),
              buf.numel() ? buf.data_ptr<float>() : nullptr, (float)lr,
              (float)momentum, (float)wd, p.numel(), sh, cur_stream());
}

void sgd_step_multi(const Tensor& table, int64_t nchunks, double lr,
                    double momentum, double wd, bool zero_grad) {
  TORCH_CHECK(table.is_cuda() && table.scalar_type() == torch::kInt64 &&
              table.is_contiguous());
  TORCH_CHECK(table.numel() >= nchunks * 6, "chunk table too small");
  al_sgd_step_multi(table.data_ptr<int64_t>(), (int)nchunks, (float)lr,
                    (float)momentum, (float)wd, zero_grad ? 1 : 0, cur_stream());
}

void sgd_step_multi_dev(const Tensor& table, int64_t nchunks,
                        const Tensor& hyper, bool zero_grad) {
  TORCH_CHECK(table.is_cuda() && table.scalar_type() == torch::kInt64 &&
              table.is_contiguous());
  TORCH_CHECK(table.numel() >= nchunks * 6, "chunk table too small");
  TORCH_CHECK(hyper.is_cuda() && hyper.scalar_type() == torch::kFloat32 &&
              hyper.numel() >= 3);
  al_sgd_step_multi_dev(table.data_ptr<int64_t>(), (int)nchunks,
                        hyper.data_ptr<float>(), zero_grad ? 1 : 0, cur_stream());
}

void wt_refresh_multi(const Tensor& table, int64_t nrows) {
  // one 64x64 (k, c) tile per row: batched (K,R,S,C)->(C,R,S,K) bf16 weight
  // transpose refresh after the fused SGD update (see optim.hip)
  TORCH_CHECK(table.is_cuda() && table.scalar_type() == torch::kInt64 &&
              table.is_contiguous());
  TORCH_CHECK(table.numel() >= nrows * 4, "wt table too small");
  al_wt_refresh(table.data_ptr<int64_t>(), (int)nrows, cur_stream());
}

void adam_step(Tensor& p, const Tensor& g, Tensor& m, Tensor& v, double lr, double b1,
               double b2, double eps, double wd, double bc1, double bc2) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kFloat32);
  al_adam_step(p.data_ptr<float>(), g.contiguous().data_ptr<float>(),
               m.data_ptr<float>(), v.data_ptr<float>(), (float)lr, (float)b1,
               (float)b2, (float)eps, (float)wd, (float)bc1, (float)bc2, p.numel(),
               cur_stream());
}

int64_t kcenter_greedy_dev(const Tensor& dist, Tensor& min_dist, Tensor& labeled,
                           Tensor& sel, const Tensor& randu, int64_t j_init,
                           bool randomize) {
  TORCH_CHECK(dist.is_cuda() && dist.scalar_type() == torch::kFloat32 &&
              dist.is_contiguous());
  const long n = dist.size(0);
  TORCH_CHECK(dist.size(1) == n && min_dist.numel() == n);
  TORCH_CHECK(labeled.scalar_type() == torch::kUInt8 && labeled.numel() == n);
  TORCH_CHECK(sel.scalar_type() == torch::kInt64);
  const int iters = (int)sel.numel();
  if (iters == 0) return 0;
  if (randomize) TORCH_CHECK(randu.numel() >= iters);
  int nblocks = (int)std::min<long>(512, (n + 1023) / 1024);
  if (nblocks < 1) nblocks = 1;
  auto fopts = dist.options();
  Tensor partial = torch::zeros({2 * nblocks}, fopts);
  Tensor st = torch::zeros({8}, fopts.dtype(torch::kInt32));
  return al_kcenter_greedy(
      dist.data_ptr<float>(), min_dist.data_ptr<float>(),
      labeled.data_ptr<unsigned char>(), sel.data_ptr<int64_t>(),
      randomize ? randu.data_ptr<float>() : nullptr, partial.data_ptr<float>(),
      st.data_ptr(), n, iters, (int)j_init, randomize ? 1 : 0, nblocks,
      cur_stream());
}

Tensor scatter_s2(const Tensor& tmp, int64_t H, int64_t W, int64_t stride) {
  check_bf16_contig(tmp, "tmp");
  const int N = (int)tmp.size(0), P = (int)tmp.size(1), Q = (int)tmp.size(2),
            C = (int)tmp.size(3);
  TORCH_CHECK(C % 8 == 0);
  Tensor dx = torch::empty({(long)N, H, W, (long)C}, tmp.options());
  al_scatter_s2(tmp.data_ptr(), dx.data_ptr(), N, (int)H, (int)W, C, P, Q,
                (int)stride, cur_stream());
  return dx;
}

Tensor pairwise_sqdist_dev(const Tensor& f, const Tensor& sq) {
  check_bf16_contig(f, "features");
  TORCH_CHECK(sq.scalar_type() == torch::kFloat32 && sq.is_contiguous());
  const long N = f.size(0);
  const int M = (int)f.size(1);
  TORCH_CHECK(sq.numel() == N && M % 64 == 0);
  Tensor out = torch::empty({N, N}, sq.options());
  int rc = al_pairwise_sqdist(f.data_ptr(), sq.data_ptr<float>(),
                              out.data_ptr<float>(), zero_page(f).data_ptr(),
                              N, M, cur_stream());
  TORCH_CHECK(rc == 0, "pairwise_sqdist: M % 64 != 0");
  return out;
}

Tensor badge_gram(const Tensor& a, const Tensor& e, const Tensor& d) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kFloat32 && a.is_contiguous());
  TORCH_CHECK(e.scalar_type() == torch::kFloat32 && e.is_contiguous());
  TORCH_CHECK(d.scalar_type() == torch::kFloat32 && d.is_contiguous());
  const long N = a.size(0);
  TORCH_CHECK(e.size(0) == N && d.numel() == N);
  Tensor out = torch::empty({N, N}, a.options());
  al_badge_gram(a.data_ptr<float>(), e.data_ptr<float>(), d.data_ptr<float>(),
                out.data_ptr<float>(), N, (int)a.size(1), (int)e.size(1),
                cur_stream());
  return out;
}

Tensor linear_fwd(const Tensor& x, const Tensor& w, const Tensor& bias) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kFloat32 && x.is_contiguous());
  TORCH_CHECK(w.scalar_type() == torch::kFloat32 && w.is_contiguous());
  const int B = (int)x.size(0), M = (int)x.size(1), C = (int)w.size(0);
  TORCH_CHECK(w.size(1) == M);
  Tensor out = al_linear_needs_zero(B, C, M)
                   ? torch::zeros({B, C}, x.options())
                   : torch::empty({B, C}, x.options());
  al_linear_fwd(x.data_ptr<float>(), w.data_ptr<float>(),
                bias.numel() ? bias.data_ptr<float>() : nullptr,
                out.data_ptr<float>(), B, M, C, cur_stream());
  return out;
}

std::vector<Tensor> linear_bwd(const Tensor& dy, const Tensor& x, const Tensor& w,
                               bool want_dx, bool want_dw, bool want_db) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kFloat32 &&
              dy.is_contiguous());
  const int B = (int)x.size(0), M = (int)x.size(1), C = (int)w.size(0);
  auto alloc = [&](long r, long c, int K_) {
    return al_linear_needs_zero((int)r, (int)c, K_)
               ? torch::zeros({r, c}, x.options())
               : torch::empty({r, c}, x.options());
  };
  Tensor dx = want_dx ? alloc(B, M, C) : Tensor();
  Tensor dw = want_dw ? alloc(C, M, B) : Tensor();
  Tensor db = want_db ? torch::empty({C}, x.options()) : Tensor();
  al_linear_bwd(dy.data_ptr<float>(), x.contiguous().data_ptr<float>(),
                w.contiguous().data_ptr<float>(),
                want_dx ? dx.data_ptr<float>() : nullptr,
                want_dw ? dw.data_ptr<float>() : nullptr,
                want_db ? db.data_ptr<float>() : nullptr, B, M, C, cur_stream());
  return {dx, dw, db};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv2d_fwd", &conv2d_fwd);
  m.def("conv2d_fwd_fused", &conv2d_fwd_fused);
  m.def("conv2d_fwd_stats", &conv2d_fwd_stats);
  m.def("conv2d_bwd_data", &conv2d_bwd_data);
  m.def("conv2d_bwd_data_bn", &conv2d_bwd_data_bn);
  m.def("conv2d_bwd_data_res", &conv2d_bwd_data_res);
  m.def("conv2d_bwd_weight", &conv2d_bwd_weight);
  m.def("conv2d_bwd_weight_into", &conv2d_bwd_weight_into);
  m.def("bn_stats", &bn_stats);
  m.def("bn_stats_finalize", &bn_stats_finalize);
  m.def("bn_finalize", &bn_finalize);
  m.def("bn_norm_fwd", &bn_norm_fwd);
  m.def("bn_bwd_reduce", &bn_bwd_reduce);
  m.def("bn_bwd", &bn_bwd);
  m.def("im2col_pack", &im2col_pack);
  m.def("maxpool2d_fwd", &maxpool2d_fwd);
  m.def("maxpool2d_bwd", &maxpool2d_bwd);
  m.def("global_avg_pool", &global_avg_pool);
  m.def("softmax_scores", &softmax_scores);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("sgd_step", &sgd_step);
  m.def("sgd_step_multi", &sgd_step_multi);
  m.def("sgd_step_multi_dev", &sgd_step_multi_dev);
  m.def("wt_refresh_multi", &wt_refresh_multi);
  m.def("kcenter_greedy_dev", &kcenter_greedy_dev);
  m.def("scatter_s2", &scatter_s2);
  m.def("badge_gram", &badge_gram);
  m.def("pairwise_sqdist_dev", &pairwise_sqdist_dev);
  m.def("linear_fwd", &linear_fwd);
  m.def("linear_bwd", &linear_bwd);
  m.def("adam_step", &adam_step);
}
