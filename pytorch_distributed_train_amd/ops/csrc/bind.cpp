// Python binding for the gfx950 kernels + the RCCL communicator.
// This is the only torch-facing translation unit; the .hip files are pure
// CDNA4 device code exposed as extern "C" launchers.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>  // ROCm-native torch GPU context

#include <hip/hip_runtime_api.h>
#include <rccl/rccl.h>

#include <vector>

// ---- extern "C" launchers (see the .hip files) ---------------------------
extern "C" {
void pdta_relu_fwd(const void*, void*, long, int, hipStream_t);
void pdta_relu_bwd_mask(const void*, const void*, void*, long, int,
                        hipStream_t);
void pdta_add_relu(const void*, const void*, void*, long, int, hipStream_t);
void pdta_bn_stats(const void*, float*, float*, float*, float*, float*,
                   float, float, int, long, int, int, hipStream_t);
int pdta_bn_chunks(long, int);
void pdta_bn_apply(const void*, void*, const void*, const float*,
                   const float*, const float*, const float*, long, int, int,
                   int, unsigned char*, hipStream_t);
void pdta_bn_bwd_reduce(const void*, const void*, const void*,
                        const float*, const float*, const float*,
                        const float*, int, float*, float*, float*, long,
                        int, int, const void*, void*, const unsigned char*,
                        hipStream_t);
void pdta_bn_bwd_elemt(const void*, const void*, const void*, void*,
                       void*, const float*, const float*, int,
                       const float*, const float*, const float*,
                       const float*, float, long, int, int, const void*,
                       hipStream_t);
void pdta_syncbn_pack(const float*, const float*, float, float*, int,
                      hipStream_t);
void pdta_syncbn_combine(const float*, int, int, float, float, float*,
                         float*, float*, float*, float*, hipStream_t);
void pdta_syncbn_pack2(const float*, const float*, float, float*, int,
                       hipStream_t);
void pdta_syncbn_div(float*, int, hipStream_t);
void pdta_maxpool_fwd(const void*, void*, uint8_t*, int, int, int, int, int,
                      int, int, int, int, int, hipStream_t);
void pdta_maxpool_bwd(const void*, const uint8_t*, void*, int, int, int, int,
                      int, int, int, int, int, int, hipStream_t);
void pdta_gap_fwd(const void*, void*, int, long, int, int, hipStream_t);
void pdta_gap_bwd(const void*, void*, int, long, int, int, hipStream_t);
void pdta_fused_sgd(float*, const float*, float*, long, float, float, float,
                    float, int, float, const float*, hipStream_t);
void pdta_unscale_check(float*, long, float, float*, hipStream_t);
void pdta_ce_fwd(const void*, const long*, float*, float*, int, int, float,
                 int, hipStream_t);
void pdta_ce_bwd(const void*, const long*, const float*, const float*, void*,
                 int, int, float, float, int, hipStream_t);
void pdta_gemm_nt(const void*, const void*, void*, const float*, int, int,
                  int, int, int, hipStream_t);
int pdta_colsum_ws(int);
void pdta_colsum_f32(const void*, float*, float*, long, int, int,
                     hipStream_t);
long pdta_conv_fwd_chunks(long, int);
void pdta_bn_stats_from_part(float*, int, float*, float*, float*, float*,
                             float, float, int, long, int, hipStream_t);
void pdta_conv_fwd(const void*, const void*, void*, const void*, int, int,
                   int, int, int, int, int, int, int, int, int, int,
                   hipStream_t, float*);
void pdta_conv_bwd_data(const void*, const void*, void*, const void*,
                        const void*, int, int, int, int, int, int, int,
                        int, int, int, int, int, hipStream_t);
void pdta_wrw_plan(int, int, int, int, int, int, int, long*, long*);
void pdta_conv_bwd_weight(const void*, const void*, float*, float*,
                          const void*, int, int, int, int, int, int, int,
                          int, int, int, int, int, hipStream_t);
void pdta_stem_im2col(const void*, void*, int, int, int, int, int, int,
                      int, int, int, int, int, int, hipStream_t);
void pdta_gemm_nt_batched(const void*, const void*, void*, int, int, int,
                          int, long, long, long, int, int, hipStream_t);
void pdta_ln_fwd(const void*, void*, const float*, const float*, float*,
                 float*, long, int, float, int, const void*, void*,
                 hipStream_t);
int pdta_ln_chunks(long);
void pdta_ln_bwd(const void*, const void*, void*, const float*,
                 const float*, const float*, float*, float*, float*, long,
                 int, int, const void*, hipStream_t);
void pdta_gelu_fwd(const void*, void*, long, int, hipStream_t);
void pdta_gelu_bwd(const void*, const void*, void*, long, int, hipStream_t);
void pdta_softmax_fwd(const float*, void*, long, int, int, float, int,
                      hipStream_t);
void pdta_softmax_bwd(const void*, const void*, void*, long, int, float,
                      int, hipStream_t);
void pdta_gemm_bs(const void*, const void*, void*, int, int, int, int, int,
                  int, int, int, long, long, long, long, long, long, int,
                  int, int, int, int, int, hipStream_t);
void pdta_pack_weights(const float*, void*, const long*, const long*, int,
                       long, int, hipStream_t);
void pdta_pack_transpose(const float*, void*, const long*, const long*,
                         int, long, int, hipStream_t);
int pdta_attn_fused_ok(int, int);
void pdta_attn_fused_fwd(const void*, void*, void*, int, int, int, int,
                         float, int, hipStream_t);
void pdta_attn_fused_bwd_ds(const void*, const void*, const void*, void*,
                            void*, int, int, int, int, float, int,
                            hipStream_t);
}

namespace {

int dt_of(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return 0;
    case at::kBFloat16: return 1;
    case at::kHalf: return 2;
    default: TORCH_CHECK(false, "unsupported dtype ", t.scalar_type());
  }
}

hipStream_t cur_stream() {
  return (hipStream_t)at::hip::getCurrentHIPStreamMasqueradingAsCUDA()
      .stream();
}

#define CHECK_IN(t)                                            \
  TORCH_CHECK(t.is_cuda(), #t " must be on GPU");              \
  TORCH_CHECK(t.is_contiguous(), #t " must be contiguous")

// 256-B zero page for out-of-bounds im2col taps (per device, lazy)
const void* zero_page(const at::Tensor& like) {
  static at::Tensor zp;
  if (!zp.defined() || zp.device() != like.device())
    zp = at::zeros({256}, like.options().dtype(at::kByte));
  return zp.data_ptr();
}

int conv_out(int h, int k, int stride, int pad) {
  return (h + 2 * pad - k) / stride + 1;
}

// ---- elementwise ---------------------------------------------------------
at::Tensor relu_fwd(at::Tensor x) {
  CHECK_IN(x);
  auto y = at::empty_like(x);
  pdta_relu_fwd(x.data_ptr(), y.data_ptr(), x.numel(), dt_of(x),
                cur_stream());
  return y;
}

at::Tensor relu_bwd_mask(at::Tensor dy, at::Tensor y) {
  CHECK_IN(dy);
  CHECK_IN(y);
  auto dx = at::empty_like(dy);
  pdta_relu_bwd_mask(dy.data_ptr(), y.data_ptr(), dx.data_ptr(), dy.numel(),
                     dt_of(dy), cur_stream());
  return dx;
}

at::Tensor add_relu_fwd(at::Tensor a, at::Tensor b) {
  CHECK_IN(a);
  CHECK_IN(b);
  auto y = at::empty_like(a);
  pdta_add_relu(a.data_ptr(), b.data_ptr(), y.data_ptr(), a.numel(),
                dt_of(a), cur_stream());
  return y;
}

// ---- batchnorm -----------------------------------------------------------
std::vector<at::Tensor> bn_stats(at::Tensor x) {
  // (mean, var) — SyncBN partials / parity path (no running update)
  CHECK_IN(x);
  int C = x.size(3);
  long rows = x.numel() / C;
  auto opts = x.options().dtype(at::kFloat);
  int chunks = pdta_bn_chunks(rows, C);
  auto part = at::empty({(long)chunks * 2 * C}, opts);
  auto mean = at::empty({C}, opts), var = at::empty({C}, opts);
  pdta_bn_stats(x.data_ptr(), part.data_ptr<float>(), mean.data_ptr<float>(),
                var.data_ptr<float>(), nullptr, nullptr, 0.f, 0.f, 0, rows,
                C, dt_of(x), cur_stream());
  return {mean, var};
}

std::vector<at::Tensor> bn_stats_train(at::Tensor x, at::Tensor running_mean,
                                       at::Tensor running_var,
                                       double momentum, double eps) {
  // (mean, invstd); updates running stats in-kernel (training fast path)
  CHECK_IN(x);
  int C = x.size(3);
  long rows = x.numel() / C;
  auto opts = x.options().dtype(at::kFloat);
  int chunks = pdta_bn_chunks(rows, C);
  auto part = at::empty({(long)chunks * 2 * C}, opts);
  auto mean = at::empty({C}, opts), invstd = at::empty({C}, opts);
  pdta_bn_stats(x.data_ptr(), part.data_ptr<float>(), mean.data_ptr<float>(),
                invstd.data_ptr<float>(), running_mean.data_ptr<float>(),
                running_var.data_ptr<float>(), momentum, eps, 1, rows, C,
                dt_of(x), cur_stream());
  return {mean, invstd};
}

std::vector<at::Tensor> bn_apply_v(at::Tensor x, at::Tensor weight,
                                   at::Tensor bias, at::Tensor mean,
                                   at::Tensor invstd, bool relu,
                                   c10::optional<at::Tensor> res,
                                   bool want_mask) {
  // want_mask (joins only): also return the 1-bit relu mask (1 byte per
  // 8 elems) the backward reduce consumes instead of re-reading y
  CHECK_IN(x);
  int C = x.size(3);
  long rows = x.numel() / C;
  auto y = at::empty_like(x);
  auto w = weight.to(at::kFloat).contiguous();
  auto b = bias.to(at::kFloat).contiguous();
  auto m = mean.contiguous(), is = invstd.contiguous();
  at::Tensor mask;
  unsigned char* mp = nullptr;
  if (want_mask) {
    TORCH_CHECK(relu && res.has_value(), "bn_apply mask: joins only");
    mask = at::empty({rows * C / 8}, x.options().dtype(at::kByte));
    mp = mask.data_ptr<unsigned char>();
  }
  pdta_bn_apply(x.data_ptr(), y.data_ptr(),
                res.has_value() ? res->data_ptr() : nullptr,
                w.data_ptr<float>(), b.data_ptr<float>(),
                m.data_ptr<float>(), is.data_ptr<float>(), rows, C,
                relu ? 1 : 0, dt_of(x), mp, cur_stream());
  if (want_mask) return {y, mask};
  return {y};
}

at::Tensor bn_apply(at::Tensor x, at::Tensor weight, at::Tensor bias,
                    at::Tensor mean, at::Tensor invstd, bool relu,
                    c10::optional<at::Tensor> res) {
  return bn_apply_v(x, weight, bias, mean, invstd, relu, res, false)[0];
}

std::vector<at::Tensor> bn_bwd_reduce(at::Tensor x, at::Tensor dy,
                                      c10::optional<at::Tensor> relu_y,
                                      at::Tensor mean,
                                      c10::optional<at::Tensor> invstd,
                                      c10::optional<at::Tensor> weight,
                                      c10::optional<at::Tensor> bias,
                                      c10::optional<at::Tensor> dy2,
                                      c10::optional<at::Tensor> mbits,
                                      c10::optional<at::Tensor> sum_dy_out) {
  // sum_dy_out: direct-grad destination (the BN bias flat slice) — the
  // finalize kernel writes dbias in place, no D2D copy afterwards
  // mask modes: relu_y given -> read it; else invstd/weight/bias given ->
  // recompute mask (gamma*xhat+beta > 0), skipping the relu_y pass;
  // else no mask
  CHECK_IN(x);
  CHECK_IN(dy);
  int C = x.size(3);
  long rows = x.numel() / C;
  auto opts = x.options().dtype(at::kFloat);
  int chunks = pdta_bn_chunks(rows, C);
  auto part = at::empty({(long)chunks * 2 * C}, opts);
  at::Tensor sum_dy = sum_dy_out.has_value()
                          ? sum_dy_out->view({C})
                          : at::empty({C}, opts);
  auto sum_dy_xmu = at::empty({C}, opts);
  int mask = relu_y.has_value() ? 1 : (bias.has_value() ? 2 : 0);
  at::Tensor dres;  // two-dy: masked combined dy, written by the kernel
  if (dy2.has_value()) dres = at::empty_like(dy);
  at::Tensor isf, wf, bf;
  if (mask == 2) {
    isf = invstd->to(at::kFloat).contiguous();
    wf = weight->to(at::kFloat).contiguous();
    bf = bias->to(at::kFloat).contiguous();
  }
  TORCH_CHECK(!dy2.has_value() || mask == 1 || mbits.has_value(),
              "bn_bwd_reduce: dy2 requires a relu mask");
  if (dy2.has_value())
    TORCH_CHECK(dy2->is_contiguous() && dy2->sizes() == dy.sizes() &&
                dy2->scalar_type() == dy.scalar_type(),
                "bn_bwd_reduce: dy2 shape/dtype mismatch");
  pdta_bn_bwd_reduce(x.data_ptr(), dy.data_ptr(),
                     relu_y.has_value() ? relu_y->data_ptr() : nullptr,
                     mean.contiguous().data_ptr<float>(),
                     mask == 2 ? isf.data_ptr<float>() : nullptr,
                     mask == 2 ? wf.data_ptr<float>() : nullptr,
                     mask == 2 ? bf.data_ptr<float>() : nullptr, mask,
                     part.data_ptr<float>(), sum_dy.data_ptr<float>(),
                     sum_dy_xmu.data_ptr<float>(), rows, C, dt_of(x),
                     dy2.has_value() ? dy2->data_ptr() : nullptr,
                     dy2.has_value() ? dres.data_ptr() : nullptr,
                     mbits.has_value() ? mbits->data_ptr<unsigned char>()
                                       : nullptr,
                     cur_stream());
  if (dy2.has_value()) return {sum_dy, sum_dy_xmu, dres};
  return {sum_dy, sum_dy_xmu};
}

std::vector<at::Tensor> bn_bwd_elemt(at::Tensor x, at::Tensor dy,
                                     c10::optional<at::Tensor> relu_y,
                                     at::Tensor mean, at::Tensor invstd,
                                     at::Tensor weight, at::Tensor sum_dy,
                                     at::Tensor sum_dy_xmu, double count,
                                     bool want_dres,
                                     c10::optional<at::Tensor> bias,
                                     c10::optional<at::Tensor> dy2) {
  CHECK_IN(x);
  CHECK_IN(dy);
  int C = x.size(3);
  long rows = x.numel() / C;
  auto dx = at::empty_like(x);
  at::Tensor dres;
  if (want_dres) dres = at::empty_like(x);
  auto w = weight.to(at::kFloat).contiguous();
  int mask = relu_y.has_value() ? 1 : (bias.has_value() ? 2 : 0);
  at::Tensor bf;
  if (mask == 2) bf = bias->to(at::kFloat).contiguous();
  TORCH_CHECK(!(want_dres && mask != 1),
              "bn_bwd_elemt: dres requires the stored-relu mask");
  TORCH_CHECK(!dy2.has_value() || mask == 1,
              "bn_bwd_elemt: dy2 requires the stored-relu mask");
  if (dy2.has_value())
    TORCH_CHECK(dy2->is_contiguous() && dy2->sizes() == dy.sizes() &&
                dy2->scalar_type() == dy.scalar_type(),
                "bn_bwd_elemt: dy2 shape/dtype mismatch");
  pdta_bn_bwd_elemt(x.data_ptr(), dy.data_ptr(),
                    relu_y.has_value() ? relu_y->data_ptr() : nullptr,
                    dx.data_ptr(), want_dres ? dres.data_ptr() : nullptr,
                    w.data_ptr<float>(),
                    mask == 2 ? bf.data_ptr<float>() : nullptr, mask,
                    mean.contiguous().data_ptr<float>(),
                    invstd.contiguous().data_ptr<float>(),
                    sum_dy.contiguous().data_ptr<float>(),
                    sum_dy_xmu.contiguous().data_ptr<float>(), (float)count,
                    rows, C, dt_of(x),
                    dy2.has_value() ? dy2->data_ptr() : nullptr,
                    cur_stream());
  if (want_dres) return {dx, dres};
  return {dx};
}


// ---- SyncBN fused pack/combine ------------------------------------------
at::Tensor syncbn_pack(at::Tensor mean, at::Tensor var, double n) {
  CHECK_IN(mean);
  int C = mean.numel();
  auto out = at::empty({2 * C + 1}, mean.options());
  pdta_syncbn_pack(mean.data_ptr<float>(), var.contiguous().data_ptr<float>(),
                   (float)n, out.data_ptr<float>(), C, cur_stream());
  return out;
}

std::vector<at::Tensor> syncbn_combine(at::Tensor gathered, int64_t ws,
                                       int64_t C, double eps,
                                       double momentum,
                                       c10::optional<at::Tensor> running_mean,
                                       c10::optional<at::Tensor> running_var) {
  CHECK_IN(gathered);
  auto opts = gathered.options();
  auto mean = at::empty({C}, opts), invstd = at::empty({C}, opts);
  auto total = at::empty({1}, opts);
  pdta_syncbn_combine(
      gathered.data_ptr<float>(), ws, C, (float)eps, (float)momentum,
      mean.data_ptr<float>(), invstd.data_ptr<float>(),
      running_mean.has_value() ? running_mean->data_ptr<float>() : nullptr,
      running_var.has_value() ? running_var->data_ptr<float>() : nullptr,
      total.data_ptr<float>(), cur_stream());
  return {mean, invstd, total};
}

at::Tensor syncbn_pack2(at::Tensor a, at::Tensor b, double n) {
  CHECK_IN(a);
  int C = a.numel();
  auto out = at::empty({2 * C + 1}, a.options());
  pdta_syncbn_pack2(a.data_ptr<float>(), b.contiguous().data_ptr<float>(),
                    (float)n, out.data_ptr<float>(), C, cur_stream());
  return out;
}

void syncbn_div(at::Tensor packed, int64_t C) {
  CHECK_IN(packed);
  pdta_syncbn_div(packed.data_ptr<float>(), C, cur_stream());
}

// ---- pooling -------------------------------------------------------------
std::vector<at::Tensor> maxpool_fwd(at::Tensor x, int64_t k, int64_t stride,
                                    int64_t pad) {
  CHECK_IN(x);
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int P = conv_out(H, k, stride, pad), Q = conv_out(W, k, stride, pad);
  auto y = at::empty({N, P, Q, C}, x.options());
  auto idx = at::empty({N, P, Q, C}, x.options().dtype(at::kByte));
  pdta_maxpool_fwd(x.data_ptr(), y.data_ptr(), idx.data_ptr<uint8_t>(), N, H,
                   W, C, P, Q, k, stride, pad, dt_of(x), cur_stream());
  return {y, idx};
}

at::Tensor maxpool_bwd(at::Tensor dy, at::Tensor idx, int64_t H, int64_t W,
                       int64_t k, int64_t stride, int64_t pad) {
  CHECK_IN(dy);
  CHECK_IN(idx);
  int N = dy.size(0), P = dy.size(1), Q = dy.size(2), C = dy.size(3);
  auto dx = at::empty({N, (long)H, (long)W, (long)C}, dy.options());
  pdta_maxpool_bwd(dy.data_ptr(), idx.data_ptr<uint8_t>(), dx.data_ptr(), N,
                   H, W, C, P, Q, k, stride, pad, dt_of(dy), cur_stream());
  return dx;
}

at::Tensor avgpool_global_fwd(at::Tensor x) {
  CHECK_IN(x);
  int N = x.size(0), C = x.size(3);
  long HW = (long)x.size(1) * x.size(2);
  auto y = at::empty({N, C}, x.options());
  pdta_gap_fwd(x.data_ptr(), y.data_ptr(), N, HW, C, dt_of(x), cur_stream());
  return y;
}

at::Tensor avgpool_global_bwd(at::Tensor dy, int64_t H, int64_t W) {
  CHECK_IN(dy);
  int N = dy.size(0), C = dy.size(1);
  auto dx = at::empty({N, H, W, C}, dy.options());
  pdta_gap_bwd(dy.data_ptr(), dx.data_ptr(), N, (long)H * W, C, dt_of(dy),
               cur_stream());
  return dx;
}

// ---- optimizer / amp -----------------------------------------------------
void fused_sgd(at::Tensor p, at::Tensor g, at::Tensor m, double lr,
               double mu, double wd, double damp, bool nesterov, double gs,
               c10::optional<at::Tensor> lr_dev) {
  CHECK_IN(p);
  CHECK_IN(g);
  pdta_fused_sgd(p.data_ptr<float>(), g.data_ptr<float>(),
                 m.numel() ? m.data_ptr<float>() : nullptr, p.numel(), lr,
                 mu, wd, damp, nesterov ? 1 : 0, gs,
                 lr_dev.has_value() ? lr_dev->data_ptr<float>() : nullptr,
                 cur_stream());
}

at::Tensor unscale_check(at::Tensor g, double inv) {
  CHECK_IN(g);
  auto found = at::zeros({1}, g.options().dtype(at::kFloat));
  pdta_unscale_check(g.data_ptr<float>(), g.numel(), inv,
                     found.data_ptr<float>(), cur_stream());
  return found;
}

// ---- loss ----------------------------------------------------------------
std::vector<at::Tensor> softmax_ce_fwd(at::Tensor logits, at::Tensor labels,
                                       double eps) {
  CHECK_IN(logits);
  CHECK_IN(labels);
  int B = logits.size(0), n = logits.size(1);
  auto opts = logits.options().dtype(at::kFloat);
  auto loss = at::empty({B}, opts), lse = at::empty({B}, opts);
  pdta_ce_fwd(logits.data_ptr(), labels.data_ptr<long>(),
              loss.data_ptr<float>(), lse.data_ptr<float>(), B, n, eps,
              dt_of(logits), cur_stream());
  return {loss, lse};
}

at::Tensor softmax_ce_bwd(at::Tensor logits, at::Tensor labels,
                          at::Tensor lse, double eps, double scale,
                          at::Tensor gout) {
  CHECK_IN(logits);
  auto d = at::empty_like(logits);
  pdta_ce_bwd(logits.data_ptr(), labels.data_ptr<long>(),
              lse.data_ptr<float>(), gout.data_ptr<float>(), d.data_ptr(),
              logits.size(0), logits.size(1), eps, scale, dt_of(logits),
              cur_stream());
  return d;
}

// ---- GEMM ----------------------------------------------------------------
at::Tensor gemm_nt(at::Tensor a, at::Tensor b,
                   c10::optional<at::Tensor> bias, bool out_f32) {
  CHECK_IN(a);
  CHECK_IN(b);
  int M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "gemm_nt: K mismatch");
  TORCH_CHECK(K % 64 == 0, "gemm_nt: K must be a multiple of 64 (pad)");
  auto y = at::empty({M, N},
                     out_f32 ? a.options().dtype(at::kFloat) : a.options());
  const float* bp = nullptr;
  at::Tensor bf;
  if (bias.has_value() && bias->numel()) {
    bf = bias->to(at::kFloat).contiguous();
    bp = bf.data_ptr<float>();
  }
  pdta_gemm_nt(a.data_ptr(), b.data_ptr(), y.data_ptr(), bp, M, N, K,
               dt_of(a), out_f32 ? 1 : 0, cur_stream());
  return y;
}

at::Tensor colsum_f32(at::Tensor dy, c10::optional<at::Tensor> out) {
  CHECK_IN(dy);
  int N = dy.size(1);
  int ws = pdta_colsum_ws(N);
  auto opts = dy.options().dtype(at::kFloat);
  // scalar fallback (ws == 0) atomics into db and needs it zeroed
  at::Tensor db = out.has_value()
                      ? (ws ? out->view({N})
                            : out->view({N}).zero_())
                      : (ws ? at::empty({N}, opts) : at::zeros({N}, opts));
  at::Tensor part;
  float* pp = nullptr;
  if (ws) {
    part = at::empty({ws}, opts);
    pp = part.data_ptr<float>();
  }
  pdta_colsum_f32(dy.data_ptr(), db.data_ptr<float>(), pp, dy.size(0), N,
                  dt_of(dy), cur_stream());
  return db;
}

// ---- conv ----------------------------------------------------------------
std::vector<at::Tensor> conv2d_fwd(at::Tensor x, at::Tensor wt,
                                   int64_t stride, int64_t pad,
                                   bool want_bn_stats, bool want_xcol) {
  // x [N,H,W,C]; wt pack [Kout,R,S,C]. want_bn_stats: the fwd kernel's
  // epilogue also emits per-m-tile (sum, sumsq) channel partials of y so
  // the following BatchNorm skips its stage-1 read of y
  // (bn_stats_train_from_part finalizes them).
  CHECK_IN(x);
  CHECK_IN(wt);
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int Kout = wt.size(0), R = wt.size(1), S = wt.size(2);
  TORCH_CHECK(wt.size(3) == C, "conv: channel mismatch");
  int P = conv_out(H, R, stride, pad), Q = conv_out(W, S, stride, pad);
  auto y = at::empty({N, P, Q, Kout}, x.options());
  at::Tensor part;
  float* part_ptr = nullptr;
  if (want_bn_stats) {
    long chunks = pdta_conv_fwd_chunks((long)N * P * Q, Kout);
    part = at::empty({(chunks + 64) * 2 * Kout},
                     x.options().dtype(at::kFloat));
    part_ptr = part.data_ptr<float>();
  }
  if (C % 64 == 0) {
    pdta_conv_fwd(x.data_ptr(), wt.data_ptr(), y.data_ptr(), zero_page(x), N,
                  H, W, C, P, Q, Kout, R, S, stride, pad, dt_of(x),
                  cur_stream(), part_ptr);
  } else {
    // stem (small C): im2col into the SAME igemm path (1x1 over xcol)
    int rsc = R * S * C, rsc_pad = (rsc + 63) / 64 * 64;
    auto xcol = at::empty({(long)N * P * Q, rsc_pad}, x.options());
    pdta_stem_im2col(x.data_ptr(), xcol.data_ptr(), N, H, W, C, P, Q, R, S,
                     stride, pad, rsc_pad, dt_of(x), cur_stream());
    auto wt_pad = at::constant_pad_nd(wt.reshape({Kout, rsc}),
                                      {0, rsc_pad - rsc}).contiguous();
    pdta_conv_fwd(xcol.data_ptr(), wt_pad.data_ptr(), y.data_ptr(),
                  zero_page(x), N, P, Q, rsc_pad, P, Q, Kout, 1, 1, 1, 0,
                  dt_of(x), cur_stream(), part_ptr);
    std::vector<at::Tensor> ret = {y};
    if (want_bn_stats) ret.push_back(part);
    if (want_xcol) ret.push_back(xcol);  // reused by conv2d_bwd_weight
    return ret;
  }
  if (want_bn_stats) return {y, part};
  return {y};
}

std::vector<at::Tensor> bn_stats_from_part(at::Tensor part, long rows,
                                            int64_t C) {
  // (mean, var) from a conv2d_fwd(want_bn_stats=True) slab — SyncBN local
  // stats without re-reading the activation
  CHECK_IN(part);
  int chunks = (int)(part.numel() / (2 * C)) - 64;
  auto opts = part.options();
  auto mean = at::empty({C}, opts), var = at::empty({C}, opts);
  pdta_bn_stats_from_part(part.data_ptr<float>(), chunks,
                          mean.data_ptr<float>(), var.data_ptr<float>(),
                          nullptr, nullptr, 0.f, 0.f, 0, rows, C,
                          cur_stream());
  return {mean, var};
}

std::vector<at::Tensor> bn_stats_train_from_part(at::Tensor part, long rows,
                                                 at::Tensor running_mean,
                                                 at::Tensor running_var,
                                                 double momentum,
                                                 double eps) {
  // (mean, invstd) from a conv2d_fwd(want_bn_stats=True) partial slab;
  // updates running stats in-kernel (mirrors bn_stats_train)
  CHECK_IN(part);
  int C = running_mean.size(0);
  int chunks = (int)(part.numel() / (2 * C)) - 64;
  auto opts = part.options();
  auto mean = at::empty({C}, opts), invstd = at::empty({C}, opts);
  pdta_bn_stats_from_part(part.data_ptr<float>(), chunks,
                          mean.data_ptr<float>(), invstd.data_ptr<float>(),
                          running_mean.data_ptr<float>(),
                          running_var.data_ptr<float>(), momentum, eps, 1,
                          rows, C, cur_stream());
  return {mean, invstd};
}

at::Tensor conv2d_bwd_data(at::Tensor dy, at::Tensor w2, int64_t H,
                           int64_t W, int64_t stride, int64_t pad,
                           c10::optional<at::Tensor> accum) {
  // dy [N,P,Q,Kout]; w2 pack [R,S,C,Kout]; accum (optional, same shape as
  // dx): residual-fork gradient folded into the store epilogue
  CHECK_IN(dy);
  CHECK_IN(w2);
  int N = dy.size(0), P = dy.size(1), Q = dy.size(2), Kout = dy.size(3);
  int R = w2.size(0), S = w2.size(1), C = w2.size(2);
  TORCH_CHECK(C % 64 == 0, "bwd_data: C must be a multiple of 64 "
                           "(the stem conv never needs dx)");
  TORCH_CHECK(Kout % 64 == 0, "bwd_data: Kout must be a multiple of 64");
  auto dx = at::empty({N, H, W, (long)C}, dy.options());
  const void* acc_ptr = nullptr;
  if (accum.has_value()) {
    TORCH_CHECK(accum->is_contiguous() &&
                accum->sizes() == dx.sizes() &&
                accum->scalar_type() == dy.scalar_type(),
                "bwd_data accum: shape/dtype mismatch");
    acc_ptr = accum->data_ptr();
  }
  pdta_conv_bwd_data(dy.data_ptr(), w2.data_ptr(), dx.data_ptr(),
                     zero_page(dy), acc_ptr, N, H, W, C, P, Q, Kout, R, S,
                     stride, pad, dt_of(dy), cur_stream());
  return dx;
}

at::Tensor conv2d_bwd_weight(at::Tensor x, at::Tensor dy, int64_t R,
                             int64_t S, int64_t stride, int64_t pad,
                             c10::optional<at::Tensor> xcol_cached,
                             c10::optional<at::Tensor> out) {
  CHECK_IN(x);
  CHECK_IN(dy);
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int P = dy.size(1), Q = dy.size(2), Kout = dy.size(3);
  if (out.has_value())
    TORCH_CHECK(out->is_contiguous() &&
                out->scalar_type() == at::kFloat &&
                out->numel() == (long)Kout * C * R * S,
                "conv2d_bwd_weight: bad out (direct-grad view)");
  if (C % 64 == 0) {
    // direct-grad: the reduce writes every element exactly once, so the
    // flat-gradient slice can be the destination (no AccumulateGrad add)
    auto dw = out.has_value()
                  ? out->view({Kout, C, R, S})
                  : at::empty({Kout, C, R, S},
                              x.options().dtype(at::kFloat));
    long tiles, chunks;
    pdta_wrw_plan(N, P, Q, Kout, C, R, S, &tiles, &chunks);
    auto part = at::empty({tiles * (chunks + 16) * 4096},
                          x.options().dtype(at::kFloat));
    pdta_conv_bwd_weight(x.data_ptr(), dy.data_ptr(), dw.data_ptr<float>(),
                         part.data_ptr<float>(), zero_page(x), N, H, W, C,
                         P, Q, Kout, R, S, stride, pad, dt_of(x),
                         cur_stream());
    return dw;
  }
  // stem: im2col + 1x1 igemm wrw, then repack [K][r][s][c] -> [K,C,R,S]
  int rsc = R * S * C, rsc_pad = (rsc + 63) / 64 * 64;
  at::Tensor xcol;
  if (xcol_cached.has_value()) {  // conv2d_fwd(want_xcol=True) output
    xcol = *xcol_cached;
    TORCH_CHECK(xcol.size(0) == (long)N * P * Q &&
                xcol.size(1) == rsc_pad, "cached xcol shape mismatch");
  } else {
    xcol = at::empty({(long)N * P * Q, rsc_pad}, x.options());
    pdta_stem_im2col(x.data_ptr(), xcol.data_ptr(), N, H, W, C, P, Q, R, S,
                     stride, pad, rsc_pad, dt_of(x), cur_stream());
  }
  auto dwcol = at::empty({Kout, rsc_pad}, x.options().dtype(at::kFloat));
  long tiles, chunks;
  pdta_wrw_plan(N, P, Q, Kout, rsc_pad, 1, 1, &tiles, &chunks);
  auto part = at::empty({tiles * (chunks + 16) * 4096},
                        x.options().dtype(at::kFloat));
  pdta_conv_bwd_weight(xcol.data_ptr(), dy.data_ptr(),
                       dwcol.data_ptr<float>(), part.data_ptr<float>(),
                       zero_page(x), N, P, Q, rsc_pad, P, Q, Kout, 1, 1, 1,
                       0, dt_of(x), cur_stream());
  auto packed = dwcol.slice(1, 0, rsc).view({Kout, R, S, C});
  if (out.has_value()) {
    out->view({Kout, C, R, S})
        .copy_(packed.permute({0, 3, 1, 2}));
    return *out;
  }
  return packed.permute({0, 3, 1, 2}).contiguous();
}

// ---- transformer ops (ViT path) -----------------------------------------
at::Tensor gemm_nt_batched(at::Tensor a, at::Tensor b, bool out_f32) {
  // a [bat, M, K], b [bat, N, K] -> [bat, M, N]
  CHECK_IN(a);
  CHECK_IN(b);
  int bat = a.size(0), M = a.size(1), K = a.size(2), N = b.size(1);
  TORCH_CHECK(b.size(0) == bat && b.size(2) == K && K % 64 == 0,
              "gemm_nt_batched shape");
  auto y = at::empty({bat, M, N},
                     out_f32 ? a.options().dtype(at::kFloat) : a.options());
  pdta_gemm_nt_batched(a.data_ptr(), b.data_ptr(), y.data_ptr(), M, N, K,
                       bat, (long)M * K, (long)N * K, (long)M * N, dt_of(a),
                       out_f32 ? 1 : 0, cur_stream());
  return y;
}

std::vector<at::Tensor> ln_fwd(at::Tensor x, at::Tensor w, at::Tensor b,
                               double eps,
                               c10::optional<at::Tensor> res) {
  // res: fused residual join — returns {ln(x+res), mean, invstd, x+res}
  CHECK_IN(x);
  int d = x.size(-1);
  long rows = x.numel() / d;
  auto y = at::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  auto mean = at::empty({rows}, opts), invstd = at::empty({rows}, opts);
  auto wf = w.to(at::kFloat).contiguous(), bf = b.to(at::kFloat).contiguous();
  at::Tensor sum;
  if (res.has_value()) {
    TORCH_CHECK(res->is_contiguous() && res->sizes() == x.sizes(),
                "ln_fwd res mismatch");
    sum = at::empty_like(x);
  }
  pdta_ln_fwd(x.data_ptr(), y.data_ptr(), wf.data_ptr<float>(),
              bf.data_ptr<float>(), mean.data_ptr<float>(),
              invstd.data_ptr<float>(), rows, d, eps, dt_of(x),
              res.has_value() ? res->data_ptr() : nullptr,
              res.has_value() ? sum.data_ptr() : nullptr, cur_stream());
  if (res.has_value()) return {y, mean, invstd, sum};
  return {y, mean, invstd};
}

std::vector<at::Tensor> ln_bwd(at::Tensor x, at::Tensor dy, at::Tensor w,
                               at::Tensor mean, at::Tensor invstd,
                               c10::optional<at::Tensor> dxadd,
                               c10::optional<at::Tensor> dw_out,
                               c10::optional<at::Tensor> db_out) {
  CHECK_IN(x);
  CHECK_IN(dy);
  int d = x.size(-1);
  long rows = x.numel() / d;
  auto dx = at::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  int chunks = pdta_ln_chunks(rows);
  // +16 rows: scratch for the kernel-side two-level finalize
  auto part = at::empty({(long)(chunks + 16) * 2 * d}, opts);
  at::Tensor dw = dw_out.has_value() ? dw_out->view({d})
                                     : at::empty({d}, opts);
  at::Tensor db = db_out.has_value() ? db_out->view({d})
                                     : at::empty({d}, opts);
  auto wf = w.to(at::kFloat).contiguous();
  if (dxadd.has_value())
    TORCH_CHECK(dxadd->is_contiguous() && dxadd->sizes() == x.sizes(),
                "ln_bwd dxadd mismatch");
  pdta_ln_bwd(x.data_ptr(), dy.data_ptr(), dx.data_ptr(),
              wf.data_ptr<float>(), mean.data_ptr<float>(),
              invstd.data_ptr<float>(), part.data_ptr<float>(),
              dw.data_ptr<float>(), db.data_ptr<float>(), rows, d, dt_of(x),
              dxadd.has_value() ? dxadd->data_ptr() : nullptr,
              cur_stream());
  return {dx, dw, db};
}

at::Tensor gelu_fwd(at::Tensor x) {
  CHECK_IN(x);
  auto y = at::empty_like(x);
  pdta_gelu_fwd(x.data_ptr(), y.data_ptr(), x.numel(), dt_of(x),
                cur_stream());
  return y;
}

at::Tensor gelu_bwd(at::Tensor x, at::Tensor dy) {
  CHECK_IN(x);
  CHECK_IN(dy);
  auto dx = at::empty_like(x);
  pdta_gelu_bwd(x.data_ptr(), dy.data_ptr(), dx.data_ptr(), x.numel(),
                dt_of(x), cur_stream());
  return dx;
}

at::Tensor attn_softmax_fwd(at::Tensor scores, int64_t nvalid, double scale,
                            at::ScalarType out_dtype) {
  CHECK_IN(scores);  // fp32 [.., n]
  int n = scores.size(-1);
  long rows = scores.numel() / n;
  auto p = at::empty(scores.sizes(), scores.options().dtype(out_dtype));
  pdta_softmax_fwd(scores.data_ptr<float>(), p.data_ptr(), rows, n, nvalid,
                   scale, out_dtype == at::kBFloat16 ? 1 : 2, cur_stream());
  return p;
}

at::Tensor attn_softmax_bwd(at::Tensor probs, at::Tensor dprobs,
                            double scale) {
  CHECK_IN(probs);
  CHECK_IN(dprobs);
  int n = probs.size(-1);
  long rows = probs.numel() / n;
  auto ds = at::empty(probs.sizes(), probs.options());
  pdta_softmax_bwd(probs.data_ptr(), dprobs.data_ptr(), ds.data_ptr(),
                   rows, n, scale, dt_of(probs), cur_stream());
  return ds;
}

// ---- zero-copy multi-head attention over the interleaved qkv ------------
// qkv [b, S, 3*h*hd] contiguous; returns (out [b, S, h*hd], p [b*h, S, S]).
std::vector<at::Tensor> attn_fwd(at::Tensor qkv, int64_t b, int64_t S,
                                 int64_t h, int64_t hd, int64_t nvalid,
                                 double scale) {
  CHECK_IN(qkv);
  int dt = dt_of(qkv);
  long d = h * hd, ld = 3 * d;
  char* q = (char*)qkv.data_ptr();
  char* k = q + d * 2;   // 16-bit elements
  char* v = q + 2 * d * 2;
  if (pdta_attn_fused_ok(S, hd)) {
    // one kernel: scores never touch HBM, P written once for backward.
    // qkv may be UNPADDED (S = 197): P is padded to SP = 256 internally
    // (tail rows/cols exactly zero) so the dK/dV GEMMs run K = SP.
    const long Sp = 256;  // attn.hip SP
    auto p = at::empty({b * h, Sp, Sp}, qkv.options());
    auto out = at::empty({b, S, d}, qkv.options());
    pdta_attn_fused_fwd(qkv.data_ptr(), p.data_ptr(), out.data_ptr(), b, h,
                        S, nvalid, (float)scale, dt, cur_stream());
    return {out, p};
  }
  auto scores = at::empty({b * h, S, S}, qkv.options().dtype(at::kFloat));
  pdta_gemm_bs(q, k, scores.data_ptr(), S, S, hd, ld, ld, S, b * h, h,
               S * ld, hd, S * ld, hd, (long)h * S * S, S * S, 0, 0, 1,
               hd - 1, hd - 1, dt, cur_stream());
  auto p = at::empty({b * h, S, S}, qkv.options());
  pdta_softmax_fwd(scores.data_ptr<float>(), p.data_ptr(), b * h * S, S,
                   nvalid, scale, dt, cur_stream());
  auto out = at::empty({b, S, d}, qkv.options());
  pdta_gemm_bs(p.data_ptr(), v, out.data_ptr(), S, hd, S, S, ld, d, b * h,
               h, (long)h * S * S, S * S, S * ld, hd, S * d, hd, 0, 1, 0,
               S - 1, S - 1, dt, cur_stream());
  return {out, p};
}

// returns d(qkv) [b, S, 3*h*hd] (written in place per head, zero copies)
at::Tensor attn_bwd(at::Tensor qkv, at::Tensor p, at::Tensor dout,
                    int64_t b, int64_t S, int64_t h, int64_t hd,
                    int64_t nvalid, double scale) {
  CHECK_IN(qkv);
  CHECK_IN(p);
  CHECK_IN(dout);
  int dt = dt_of(qkv);
  long d = h * hd, ld = 3 * d;
  char* q = (char*)qkv.data_ptr();
  char* k = q + d * 2;
  char* v = q + 2 * d * 2;
  bool fused = pdta_attn_fused_ok(S, hd);
  // fused path: p/ds are padded to SP=256 with exactly-zero tail
  // rows/cols, so dK/dV run K = Sp and the unpadded Q/dout operands'
  // clamped re-reads (kbrmax = S-1) are annihilated
  long Sp = fused ? 256 : S;
  auto ds = at::empty({b * h, Sp, Sp}, qkv.options());
  auto dqkv = at::empty({b, S, ld}, qkv.options());
  char* dq = (char*)dqkv.data_ptr();
  char* dk = dq + d * 2;
  char* dv = dq + 2 * d * 2;
  if (fused) {
    // one kernel: dP never touches HBM, dQ comes straight from the
    // in-register dS (K reuses the V LDS slab)
    pdta_attn_fused_bwd_ds(qkv.data_ptr(), p.data_ptr(), dout.data_ptr(),
                           ds.data_ptr(), dqkv.data_ptr(), b, h, S, nvalid,
                           (float)scale, dt, cur_stream());
  } else {
    auto dp = at::empty({b * h, S, S}, qkv.options());
    pdta_gemm_bs(dout.data_ptr(), v, dp.data_ptr(), S, S, hd, d, ld, S,
                 b * h, h, S * d, hd, S * ld, hd, (long)h * S * S, S * S,
                 0, 0, 0, hd - 1, hd - 1, dt, cur_stream());
    pdta_softmax_bwd(p.data_ptr(), dp.data_ptr(), ds.data_ptr(), b * h * S,
                     S, scale, dt, cur_stream());
    // dq = ds @ K
    pdta_gemm_bs(ds.data_ptr(), k, dq, S, hd, S, S, ld, ld, b * h, h,
                 (long)h * S * S, S * S, S * ld, hd, S * ld, hd, 0, 1, 0,
                 S - 1, S - 1, dt, cur_stream());
  }
  // dk = ds^T @ Q
  pdta_gemm_bs(ds.data_ptr(), q, dk, S, hd, Sp, Sp, ld, ld, b * h, h,
               (long)h * Sp * Sp, Sp * Sp, S * ld, hd, S * ld, hd, 1, 1, 0,
               Sp - 1, S - 1, dt, cur_stream());
  // dv = P^T @ dOut
  pdta_gemm_bs(p.data_ptr(), dout.data_ptr(), dv, S, hd, Sp, Sp, d, ld,
               b * h, h, (long)h * Sp * Sp, Sp * Sp, S * d, hd, S * ld, hd,
               1, 1, 0, Sp - 1, S - 1, dt, cur_stream());
  return dqkv;
}

void pack_weights(at::Tensor master, at::Tensor pool, at::Tensor desc,
                  at::Tensor prefix, int64_t total_chunks) {
  CHECK_IN(master);
  CHECK_IN(pool);
  pdta_pack_weights(master.data_ptr<float>(), pool.data_ptr(),
                    desc.data_ptr<long>(), prefix.data_ptr<long>(),
                    desc.size(0), total_chunks, dt_of(pool), cur_stream());
}

// LDS-tiled [N][K] fp32 -> [K][N] dtype transpose segments (packs.hip
// pack_transpose_k — the gather form is 16x read-amplified for large K)
void pack_transpose(at::Tensor master, at::Tensor pool, at::Tensor desc,
                    at::Tensor tprefix, int64_t total_tiles) {
  CHECK_IN(master);
  CHECK_IN(pool);
  pdta_pack_transpose(master.data_ptr<float>(), pool.data_ptr(),
                      desc.data_ptr<long>(), tprefix.data_ptr<long>(),
                      desc.size(0), total_tiles, dt_of(pool),
                      cur_stream());
}

// ---- RCCL ----------------------------------------------------------------
#define RCCL_CHECK(cmd)                                                  \
  do {                                                                   \
    ncclResult_t r = (cmd);                                              \
    TORCH_CHECK(r == ncclSuccess, "RCCL error: ", ncclGetErrorString(r)); \
  } while (0)

at::Tensor rccl_unique_id() {
  ncclUniqueId id;
  RCCL_CHECK(ncclGetUniqueId(&id));
  auto t = at::empty({(long)sizeof(ncclUniqueId)},
                     at::TensorOptions().dtype(at::kByte));
  memcpy(t.data_ptr(), &id, sizeof(id));
  return t;
}

int64_t rccl_comm_init(at::Tensor uid, int64_t rank, int64_t world) {
  TORCH_CHECK(uid.numel() >= (long)sizeof(ncclUniqueId));
  ncclUniqueId id;
  memcpy(&id, uid.contiguous().data_ptr(), sizeof(id));
  ncclComm_t comm;
  RCCL_CHECK(ncclCommInitRank(&comm, world, id, rank));
  return (int64_t)(uintptr_t)comm;
}

ncclDataType_t nccl_dt(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return ncclFloat32;
    case at::kBFloat16: return ncclBfloat16;
    case at::kHalf: return ncclFloat16;
    case at::kByte: return ncclUint8;
    case at::kLong: return ncclInt64;
    default: TORCH_CHECK(false, "rccl: unsupported dtype");
  }
}

void rccl_allreduce(at::Tensor t, int64_t comm, int64_t stream, int op) {
  CHECK_IN(t);
  RCCL_CHECK(ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                           nccl_dt(t), (ncclRedOp_t)op,
                           (ncclComm_t)(uintptr_t)comm,
                           (hipStream_t)(uintptr_t)stream));
}

void rccl_allreduce_avg(at::Tensor t, int64_t comm, int64_t stream) {
  rccl_allreduce(t, comm, stream, (int)ncclAvg);
}

void rccl_allreduce_sum(at::Tensor t, int64_t comm, int64_t stream) {
  rccl_allreduce(t, comm, stream, (int)ncclSum);
}

void rccl_broadcast(at::Tensor t, int64_t src, int64_t comm, int64_t stream) {
  CHECK_IN(t);
  RCCL_CHECK(ncclBroadcast(t.data_ptr(), t.data_ptr(), t.numel(), nccl_dt(t),
                           (int)src, (ncclComm_t)(uintptr_t)comm,
                           (hipStream_t)(uintptr_t)stream));
}

void rccl_allgather(at::Tensor inp, at::Tensor out, int64_t comm,
                    int64_t stream) {
  CHECK_IN(inp);
  CHECK_IN(out);
  RCCL_CHECK(ncclAllGather(inp.data_ptr(), out.data_ptr(), inp.numel(),
                           nccl_dt(inp), (ncclComm_t)(uintptr_t)comm,
                           (hipStream_t)(uintptr_t)stream));
}

void rccl_comm_destroy(int64_t comm) {
  ncclCommDestroy((ncclComm_t)(uintptr_t)comm);
}

void rccl_comm_abort(int64_t comm) {
  // watchdog stall path: tears down in-flight collectives so a
  // desynchronized gang member cannot wedge the GPU (ncclCommAbort
  // semantics — safe to call from a non-enqueueing thread)
  ncclCommAbort((ncclComm_t)(uintptr_t)comm);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("relu_fwd", &relu_fwd);
  m.def("relu_bwd_mask", &relu_bwd_mask);
  m.def("add_relu_fwd", &add_relu_fwd);
  m.def("bn_stats", &bn_stats);
  m.def("bn_stats_train", &bn_stats_train);
  m.def("bn_apply_v", &bn_apply_v, py::arg("x"), py::arg("weight"),
        py::arg("bias"), py::arg("mean"), py::arg("invstd"),
        py::arg("relu"), py::arg("res") = c10::nullopt,
        py::arg("want_mask") = false);
  m.def("bn_apply", &bn_apply, py::arg("x"), py::arg("weight"),
        py::arg("bias"), py::arg("mean"), py::arg("invstd"),
        py::arg("relu"), py::arg("res") = c10::nullopt);
  m.def("bn_bwd_reduce", &bn_bwd_reduce, py::arg("x"), py::arg("dy"),
        py::arg("relu_y"), py::arg("mean"),
        py::arg("invstd") = c10::nullopt, py::arg("weight") = c10::nullopt,
        py::arg("bias") = c10::nullopt, py::arg("dy2") = c10::nullopt,
        py::arg("mbits") = c10::nullopt,
        py::arg("sum_dy_out") = c10::nullopt);
  m.def("bn_bwd_elemt", &bn_bwd_elemt, py::arg("x"), py::arg("dy"),
        py::arg("relu_y"), py::arg("mean"), py::arg("invstd"),
        py::arg("weight"), py::arg("sum_dy"), py::arg("sum_dy_xmu"),
        py::arg("count"), py::arg("want_dres") = false,
        py::arg("bias") = c10::nullopt, py::arg("dy2") = c10::nullopt);
  m.def("syncbn_pack", &syncbn_pack);
  m.def("syncbn_combine", &syncbn_combine, py::arg("gathered"),
        py::arg("ws"), py::arg("C"), py::arg("eps"), py::arg("momentum"),
        py::arg("running_mean") = c10::nullopt,
        py::arg("running_var") = c10::nullopt);
  m.def("syncbn_pack2", &syncbn_pack2);
  m.def("syncbn_div", &syncbn_div);
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("avgpool_global_fwd", &avgpool_global_fwd);
  m.def("avgpool_global_bwd", &avgpool_global_bwd);
  m.def("fused_sgd", &fused_sgd, py::arg("p"), py::arg("g"),
        py::arg("m"), py::arg("lr"), py::arg("mu"), py::arg("wd"),
        py::arg("damp"), py::arg("nesterov"), py::arg("gs"),
        py::arg("lr_dev") = py::none());
  m.def("unscale_check", &unscale_check);
  m.def("softmax_ce_fwd", &softmax_ce_fwd);
  m.def("softmax_ce_bwd", &softmax_ce_bwd);
  m.def("gemm_nt", &gemm_nt, py::arg("a"), py::arg("b"),
        py::arg("bias") = c10::nullopt, py::arg("out_f32") = false);
  m.def("colsum_f32", &colsum_f32, py::arg("dy"),
        py::arg("out") = py::none());
  m.def("conv2d_fwd", &conv2d_fwd, py::arg("x"), py::arg("wt"),
        py::arg("stride"), py::arg("pad"),
        py::arg("want_bn_stats") = false, py::arg("want_xcol") = false);
  m.def("bn_stats_train_from_part", &bn_stats_train_from_part);
  m.def("bn_stats_from_part", &bn_stats_from_part);
  m.def("conv2d_bwd_data", &conv2d_bwd_data, py::arg("dy"),
        py::arg("w2"), py::arg("H"), py::arg("W"),
        py::arg("stride"), py::arg("pad"),
        py::arg("accum") = py::none());
  m.def("conv2d_bwd_weight", &conv2d_bwd_weight, py::arg("x"),
        py::arg("dy"), py::arg("R"), py::arg("S"),
        py::arg("stride"), py::arg("pad"),
        py::arg("xcol_cached") = c10::nullopt,
        py::arg("out") = c10::nullopt);
  m.def("gemm_nt_batched", &gemm_nt_batched, py::arg("a"), py::arg("b"),
        py::arg("out_f32") = false);
  m.def("ln_fwd", &ln_fwd, py::arg("x"), py::arg("w"), py::arg("b"),
        py::arg("eps"), py::arg("res") = c10::nullopt);
  m.def("ln_bwd", &ln_bwd, py::arg("x"), py::arg("dy"), py::arg("w"),
        py::arg("mean"), py::arg("invstd"),
        py::arg("dxadd") = c10::nullopt,
        py::arg("dw_out") = c10::nullopt,
        py::arg("db_out") = c10::nullopt);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("attn_softmax_fwd", &attn_softmax_fwd);
  m.def("attn_softmax_bwd", &attn_softmax_bwd);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("pack_weights", &pack_weights);
  m.def("pack_transpose", &pack_transpose);
  m.def("rccl_unique_id", &rccl_unique_id);
  m.def("rccl_comm_init", &rccl_comm_init);
  m.def("rccl_allreduce_avg", &rccl_allreduce_avg);
  m.def("rccl_allreduce_sum", &rccl_allreduce_sum);
  m.def("rccl_broadcast", &rccl_broadcast);
  m.def("rccl_allgather", &rccl_allgather);
  m.def("rccl_comm_destroy", &rccl_comm_destroy);
  m.def("rccl_comm_abort", &rccl_comm_abort);
}
