// Python bindings for the colearn gfx950 kernels.
//
// Layout policy (MI355X-first, no CUDA-compat):
//  * GEMM family: bf16, row-major; NN/TN flavors route through the NT
//    kernel with an explicit transpose pass (v1 — see gemm_bf16.hip).
//  * conv family: channels-last (NHWC) activations; weights re-packed
//    K-major per call ([CO][R][S][CI] fwd, [CI][R][S][CO] dgrad).
//  * norms: LayerNorm on [.., H] bf16; BatchNorm on channels-last bf16.
//  * optimizers / DP / top-k: flat fp32 buffers.

#include <torch/extension.h>

#include <cstdlib>

#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "api.h"

#define HIP_CHECK_LAST()                                          \
  do {                                                            \
    hipError_t e_ = hipGetLastError();                            \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel launch failed: ",   \
                hipGetErrorString(e_));                           \
  } while (0)

namespace colearn {


namespace {

hipStream_t stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32, name,
              " must be a float32 GPU tensor");
}
void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16, name,
              " must be a bfloat16 GPU tensor");
}

// ---- optimizers -----------------------------------------------------------
void fused_sgd(torch::Tensor w, torch::Tensor g, torch::Tensor m,
               torch::Tensor ref, double lr, double momentum, double wd,
               double mu) {
  check_f32(w, "w");
  check_f32(g, "g");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(w.device());
  const bool has_mom = m.numel() > 0;
  const bool has_prox = ref.numel() > 0;
  const int64_t period =
      (has_prox && ref.numel() != w.numel()) ? ref.numel() : 0;
  if (period) {
    TORCH_CHECK(w.numel() % period == 0 && period % 4 == 0,
                "stacked FedProx ref must divide w and be 4-aligned");
  }
  launch_fused_sgd(w.data_ptr<float>(), g.data_ptr<float>(),
                   has_mom ? m.data_ptr<float>() : nullptr,
                   has_prox ? ref.data_ptr<float>() : nullptr, w.numel(),
                   period, static_cast<float>(lr),
                   static_cast<float>(momentum), static_cast<float>(wd),
                   static_cast<float>(mu), has_mom, has_prox, stream());
  HIP_CHECK_LAST();
}

void fused_adam(torch::Tensor w, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, torch::Tensor ref, torch::Tensor bc,
                double lr, double b1, double b2, double eps, double wd,
                double mu, double bc1, double bc2) {
  check_f32(w, "w");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(w.device());
  const bool has_prox = ref.numel() > 0;
  const int64_t period =
      (has_prox && ref.numel() != w.numel()) ? ref.numel() : 0;
  launch_fused_adam(
      w.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
      v.data_ptr<float>(), has_prox ? ref.data_ptr<float>() : nullptr,
      bc.numel() > 0 ? bc.data_ptr<float>() : nullptr, w.numel(), period,
      static_cast<float>(lr), static_cast<float>(b1),
      static_cast<float>(b2), static_cast<float>(eps),
      static_cast<float>(wd), static_cast<float>(mu),
      static_cast<float>(bc1), static_cast<float>(bc2), has_prox, stream());
  HIP_CHECK_LAST();
}

// ---- DP / top-k -----------------------------------------------------------
void sq_norm_per_row(torch::Tensor g, torch::Tensor sq) {
  const bool bf16_in = g.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16_in || g.scalar_type() == torch::kFloat32,
              "g must be fp32 or bf16");
  check_f32(sq, "sq");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(g.device());
  auto gc = g.contiguous();
  const int64_t rows = gc.size(0), cols = gc.size(1);
  int chunks = static_cast<int>(
      std::min<int64_t>(std::max<int64_t>(2048 / rows, 1), 64));
  auto partials = torch::empty({rows * chunks},
                               g.options().dtype(torch::kFloat32));
  launch_sq_norm_per_row(gc.data_ptr(), bf16_in, sq.data_ptr<float>(),
                         partials.data_ptr<float>(), rows, cols, chunks,
                         stream());
  HIP_CHECK_LAST();
}

void scale_rows_mean(torch::Tensor g, torch::Tensor scale,
                     torch::Tensor out, int64_t groups) {
  const bool bf16_in = g.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16_in || g.scalar_type() == torch::kFloat32,
              "g must be fp32 or bf16");
  TORCH_CHECK(g.size(0) % groups == 0, "rows must divide by groups");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(g.device());
  auto gc = g.contiguous();
  launch_scale_rows_mean(gc.data_ptr(), bf16_in,
                         scale.data_ptr<float>(),
                         out.data_ptr<float>(), gc.size(0), gc.size(1),
                         static_cast<int>(groups), stream());
  HIP_CHECK_LAST();
}

// ---- fused shadow sync (flat cast + per-conv dgrad-layout transposes) ----
void shadow_sync(torch::Tensor flat_w, torch::Tensor shadow,
                 torch::Tensor shadow_t, torch::Tensor segA,
                 torch::Tensor segB, int64_t transpose_blocks) {
  check_f32(flat_w, "flat_w");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(flat_w.device());
  const int nseg = segA.numel() ? static_cast<int>(segA.size(0)) : 0;
  launch_shadow_sync(
      flat_w.data_ptr<float>(), shadow.data_ptr(), flat_w.numel(),
      nseg ? segA.data_ptr<int64_t>() : nullptr,
      nseg ? segB.data_ptr<int>() : nullptr, nseg,
      nseg ? shadow_t.data_ptr() : nullptr, transpose_blocks, stream());
  HIP_CHECK_LAST();
}

// dgrad with a PRE-TRANSPOSED weight ([G][CI][R][S][CO] bf16, produced by
// shadow_sync once per step) — no per-call permute+contiguous copy.
torch::Tensor conv2d_dgrad_km_pre(torch::Tensor gy, torch::Tensor wt,
                                  int64_t H, int64_t W, int64_t stride,
                                  int64_t pad) {
  check_bf16(gy, "gy");
  check_bf16(wt, "wt");
  TORCH_CHECK(wt.is_contiguous(), "wt must be contiguous");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(gy.device());
  const int64_t G = wt.dim() == 5 ? wt.size(0) : 1;
  const int wd = wt.dim();
  ConvShape cs;
  cs.N = static_cast<int>(gy.size(0) / G);
  cs.K = static_cast<int>(gy.size(1));
  cs.P = static_cast<int>(gy.size(2));
  cs.Q = static_cast<int>(gy.size(3));
  cs.C = static_cast<int>(wt.size(wd - 4));  // [*, CI, R, S, CO]
  cs.R = static_cast<int>(wt.size(wd - 3));
  cs.S = static_cast<int>(wt.size(wd - 2));
  cs.H = static_cast<int>(H);
  cs.W = static_cast<int>(W);
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  conv_shape_fill_magic(cs);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto gx = torch::empty({gy.size(0), cs.C, cs.H, cs.W},
                         gy.options().memory_format(
                             torch::MemoryFormat::ChannelsLast));
  launch_conv_dgrad(gc.data_ptr(), wt.data_ptr(), gx.data_ptr(), cs,
                    static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  return gx;
}

// ---- fused classifier head (avgpool + per-client FC + softmax CE) --------
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> head_fwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor b, torch::Tensor y,
    int64_t G) {
  check_bf16(x, "x");
  check_f32(w, "w");
  check_f32(b, "b");
  TORCH_CHECK(x.dim() == 4 &&
                  x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "x must be channels-last NCHW");
  TORCH_CHECK(y.scalar_type() == torch::kInt64, "y must be int64");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  const int N = x.size(0), C = x.size(1);
  const int HW = x.size(2) * x.size(3);
  const int K = w.size(1);
  TORCH_CHECK(K <= 32, "head supports <=32 classes");
  TORCH_CHECK(N % G == 0, "N must be divisible by G");
  const int B = N / static_cast<int>(G);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto pooled = torch::empty({N, C}, f32);
  auto probs = torch::empty({N, K}, f32);
  auto loss_n = torch::empty({N}, f32);
  auto loss_vec = torch::empty({G}, f32);
  launch_head_fwd(x.data_ptr(), w.data_ptr<float>(), b.data_ptr<float>(),
                  y.data_ptr<int64_t>(), pooled.data_ptr<float>(),
                  probs.data_ptr<float>(), loss_n.data_ptr<float>(),
                  loss_vec.data_ptr<float>(), N, C, HW,
                  K, B, static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  return {loss_vec, pooled, probs};
}

torch::Tensor head_bwd(torch::Tensor probs, torch::Tensor pooled,
                       torch::Tensor w, torch::Tensor y,
                       torch::Tensor gloss, torch::Tensor gw,
                       torch::Tensor gb, int64_t H, int64_t W) {
  check_f32(probs, "probs");
  check_f32(pooled, "pooled");
  check_f32(w, "w");
  check_f32(gloss, "gloss");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(probs.device());
  const int N = pooled.size(0), C = pooled.size(1);
  const int K = w.size(1), G = w.size(0);
  const int HW = static_cast<int>(H * W);
  const int B = N / G;
  auto dlogits = torch::empty({N, K},
                              probs.options().dtype(torch::kFloat32));
  auto dx = torch::empty({N, C, H, W},
                         probs.options().dtype(torch::kBFloat16),
                         at::MemoryFormat::ChannelsLast);
  launch_head_bwd(probs.data_ptr<float>(), pooled.data_ptr<float>(),
                  w.data_ptr<float>(), y.data_ptr<int64_t>(),
                  gloss.data_ptr<float>(), dlogits.data_ptr<float>(),
                  dx.data_ptr(), gw.data_ptr<float>(),
                  gb.data_ptr<float>(), N, C, HW, K, B, G, stream());
  HIP_CHECK_LAST();
  return dx;
}

// ---- fast elementwise (BERT path) ----------------------------------------
torch::Tensor gelu_fwd(torch::Tensor x) {
  check_bf16(x, "x");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto xc = x.contiguous();
  auto y = torch::empty_like(xc);
  launch_gelu_fwd(xc.data_ptr(), y.data_ptr(), xc.numel(), stream());
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor gelu_bwd(torch::Tensor gy, torch::Tensor x) {
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto gc = gy.contiguous(), xc = x.contiguous();
  auto gx = torch::empty_like(gc);
  launch_gelu_bwd(gc.data_ptr(), xc.data_ptr(), gx.data_ptr(),
                  gc.numel(), stream());
  HIP_CHECK_LAST();
  return gx;
}

// Stacked GroupNorm fwd/bwd: per-sample stats, per-client affine
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> sgn_fwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor b,
    torch::Tensor residual, int64_t groups, int64_t B_per_client,
    double eps, bool relu) {
  check_bf16(x, "x");
  check_f32(w, "w");
  check_f32(b, "b");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  const bool has_res = residual.numel() > 0;
  torch::Tensor rc;
  if (has_res) rc = residual.contiguous(torch::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1);
  const int HW = x.size(2) * x.size(3);
  TORCH_CHECK(((C <= 256 && 256 % C == 0) || C % 256 == 0)
                  && groups <= 64 && C % groups == 0,
              "sgn: C must divide/be divisible by 256");
  auto y = torch::empty_like(xc);
  auto mean = torch::empty({N, groups},
                           x.options().dtype(torch::kFloat32));
  auto invstd = torch::empty_like(mean);
  launch_sgn_fwd(xc.data_ptr(), w.contiguous().data_ptr<float>(),
                 b.contiguous().data_ptr<float>(),
                 has_res ? rc.data_ptr() : nullptr, y.data_ptr(),
                 mean.data_ptr<float>(), invstd.data_ptr<float>(), N, C,
                 HW, static_cast<int>(groups),
                 static_cast<int>(B_per_client), static_cast<float>(eps),
                 relu, stream());
  HIP_CHECK_LAST();
  return {y, mean, invstd};
}

std::tuple<torch::Tensor, torch::Tensor> sgn_bwd(
    torch::Tensor gy, torch::Tensor x, torch::Tensor yout,
    torch::Tensor w, torch::Tensor mean, torch::Tensor invstd,
    int64_t groups, int64_t B_per_client, bool relu, bool need_gres) {
  check_bf16(gy, "gy");
  check_bf16(x, "x");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1);
  const int HW = x.size(2) * x.size(3);
  auto gx = torch::empty_like(xc);
  torch::Tensor gres;
  if (need_gres)
    gres = torch::empty_like(gc);
  else
    gres = torch::empty({0}, gc.options());
  launch_sgn_bwd(gc.data_ptr(), xc.data_ptr(), yout.data_ptr(),
                 w.contiguous().data_ptr<float>(),
                 mean.data_ptr<float>(), invstd.data_ptr<float>(),
                 gx.data_ptr(), need_gres ? gres.data_ptr() : nullptr, N,
                 C, HW, static_cast<int>(groups),
                 static_cast<int>(B_per_client), relu, stream());
  HIP_CHECK_LAST();
  return {gx, gres};
}

// GroupNorm per-sample param grads: x/gy channels-last bf16 [N,C,H,W]
// -> (gw, gb) fp32 [N, C]
std::tuple<torch::Tensor, torch::Tensor> gn_grad_sample(
    torch::Tensor x, torch::Tensor gy, torch::Tensor yout,
    int64_t groups, double eps, bool relu) {
  check_bf16(x, "x");
  check_bf16(gy, "gy");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  const int N = x.size(0), C = x.size(1);
  const int HW = x.size(2) * x.size(3);
  TORCH_CHECK(((C <= 256 && 256 % C == 0) || C % 256 == 0)
                  && groups <= 64 && C % groups == 0,
              "gn_grad_sample: C must divide/be divisible by 256");
  torch::Tensor yc;
  if (relu) yc = yout.contiguous(torch::MemoryFormat::ChannelsLast);
  auto gw = torch::empty({N, C}, x.options().dtype(torch::kFloat32));
  auto gb = torch::empty_like(gw);
  launch_gn_grad_sample(xc.data_ptr(), gc.data_ptr(),
                        relu ? yc.data_ptr() : nullptr,
                        gw.data_ptr<float>(), gb.data_ptr<float>(), N, C,
                        HW, static_cast<int>(groups),
                        static_cast<float>(eps), relu, stream());
  HIP_CHECK_LAST();
  return {gw, gb};
}

// column sums over dim 1 of [B, M, N] bf16 -> fp32 written into ``out``
// [B, N] (stacked bias grads, deterministic fixed-order chunks)
void colsum_f32(torch::Tensor x, torch::Tensor out) {
  check_bf16(x, "x");
  check_f32(out, "out");
  TORCH_CHECK(x.dim() == 3 && out.is_contiguous(), "colsum shapes");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto xc = x.contiguous();
  const int64_t B = x.size(0), M = x.size(1), N = x.size(2);
  // enough chunks to fill the chip: [B, chunks, N/256] blocks
  int chunks = static_cast<int>(
      std::min<int64_t>(std::max<int64_t>(4096 / ((N / 256 + 1) * B), 16),
                        128));
  auto partials = torch::empty({B * chunks * N},
                               out.options());
  launch_colsum(xc.data_ptr(), partials.data_ptr<float>(),
                out.data_ptr<float>(), B, M, N, chunks, stream());
  HIP_CHECK_LAST();
}

// ---- fused attention (seq<=128, dh=64) -----------------------------------
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> attn_fwd(
    torch::Tensor qkv, int64_t H, double scale, double p, int64_t seed) {
  check_bf16(qkv, "qkv");
  TORCH_CHECK(qkv.dim() == 5 && qkv.size(2) == 3 && qkv.size(4) == 64,
              "qkv must be [N, S, 3, H, 64]");
  const int N = qkv.size(0), S = qkv.size(1);
  TORCH_CHECK(S % 32 == 0 && S <= 128, "seq must be <=128, mult of 32");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(qkv.device());
  auto qc = qkv.contiguous();
  auto ctx = torch::empty({N, S, H * 64}, qkv.options());
  auto P = torch::empty({static_cast<int64_t>(N) * H, S, S},
                        qkv.options());
  auto mask = torch::empty({static_cast<int64_t>(N) * H * 256},
                           qkv.options().dtype(torch::kInt64));
  launch_attn_fwd(qc.data_ptr(), ctx.data_ptr(), P.data_ptr(),
                  reinterpret_cast<unsigned long long*>(
                      mask.data_ptr<int64_t>()),
                  N, S, static_cast<int>(H), static_cast<float>(scale),
                  static_cast<float>(p),
                  static_cast<unsigned long long>(seed), stream());
  HIP_CHECK_LAST();
  return {ctx, P, mask};
}

torch::Tensor attn_bwd(torch::Tensor dctx, torch::Tensor qkv,
                       torch::Tensor P, torch::Tensor mask, int64_t H,
                       double scale, double p, int64_t seed) {
  check_bf16(dctx, "dctx");
  check_bf16(qkv, "qkv");
  const int N = qkv.size(0), S = qkv.size(1);
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(qkv.device());
  auto dc = dctx.contiguous();
  auto qc = qkv.contiguous();
  auto dS = torch::empty_like(P);
  auto dqkv = torch::empty_like(qc);
  launch_attn_bwd(dc.data_ptr(), qc.data_ptr(), P.data_ptr(),
                  reinterpret_cast<const unsigned long long*>(
                      mask.data_ptr<int64_t>()),
                  dS.data_ptr(), dqkv.data_ptr(), N, S,
                  static_cast<int>(H), static_cast<float>(scale),
                  static_cast<float>(p),
                  static_cast<unsigned long long>(seed), stream());
  HIP_CHECK_LAST();
  return dqkv;
}

void philox_gaussian_add(torch::Tensor flat, int64_t seed, double std) {
  check_f32(flat, "flat");
  TORCH_CHECK(flat.is_contiguous(), "flat must be contiguous");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(flat.device());
  launch_philox_gaussian_add(flat.data_ptr<float>(), flat.numel(),
                             static_cast<unsigned long long>(seed),
                             static_cast<float>(std), stream());
  HIP_CHECK_LAST();
}

std::tuple<torch::Tensor, torch::Tensor> topk_compress(torch::Tensor flat,
                                                       int64_t k) {
  check_f32(flat, "flat");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(flat.device());
  auto fc = flat.contiguous();
  k = std::min<int64_t>(k, fc.numel());
  auto idx = torch::empty({k}, fc.options().dtype(torch::kInt32));
  auto val = torch::empty({k}, fc.options());
  auto ws = torch::zeros({1036}, fc.options().dtype(torch::kInt32));
  const int cand_cap = static_cast<int>(k) + 4096;
  auto cand = torch::empty({cand_cap}, fc.options().dtype(torch::kInt32));
  launch_topk(fc.data_ptr<float>(), fc.numel(), k, idx.data_ptr<int>(),
              val.data_ptr<float>(),
              reinterpret_cast<unsigned*>(ws.data_ptr<int>()),
              cand.data_ptr<int>(), cand_cap, stream());
  HIP_CHECK_LAST();
  return {idx, val};
}

void scatter_add_pairs(torch::Tensor out, torch::Tensor idx,
                       torch::Tensor val) {
  check_f32(out, "out");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(out.device());
  launch_scatter_add_pairs(out.data_ptr<float>(), idx.data_ptr<int>(),
                           val.data_ptr<float>(), idx.numel(), stream());
  HIP_CHECK_LAST();
}

// ---- LayerNorm ------------------------------------------------------------
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> layernorm_fwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor b, double eps,
    int64_t G) {
  check_bf16(x, "x");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto xc = x.contiguous();
  const int H = static_cast<int>(x.size(-1));
  const int64_t rows = x.numel() / H;
  TORCH_CHECK(w.numel() == G * H && rows % G == 0,
              "grouped LN: weight [G*H], rows % G == 0");
  auto y = torch::empty_like(xc);
  auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto invstd = torch::empty_like(mean);
  launch_layernorm_fwd(
      xc.data_ptr(), w.data_ptr<float>(),
      b.data_ptr<float>(), y.data_ptr(),
      mean.data_ptr<float>(), invstd.data_ptr<float>(), rows, H,
      static_cast<float>(eps), static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  auto stat_shape = x.sizes().vec();
  stat_shape.pop_back();
  return {y, mean.view(stat_shape), invstd.view(stat_shape)};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> layernorm_bwd(
    torch::Tensor gy, torch::Tensor x, torch::Tensor w, torch::Tensor mean,
    torch::Tensor invstd, int64_t G) {
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto gc = gy.contiguous(), xc = x.contiguous();
  auto mc = mean.contiguous(), ic = invstd.contiguous();
  const int H = static_cast<int>(x.size(-1));
  const int64_t rows = x.numel() / H;
  auto gx = torch::empty_like(xc);
  auto gw_gb = torch::empty({G, 2 * H},
                            x.options().dtype(torch::kFloat32));
  auto partials = torch::empty({G * 64 * 2 * H},
                               x.options().dtype(torch::kFloat32));
  launch_layernorm_bwd(
      gc.data_ptr(), xc.data_ptr(), w.data_ptr<float>(),
      mc.data_ptr<float>(), ic.data_ptr<float>(), gx.data_ptr(),
      gw_gb.data_ptr<float>(), partials.data_ptr<float>(), rows, H,
      static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  return {gx, gw_gb.narrow(1, 0, H).clone(),
          gw_gb.narrow(1, H, H).clone()};
}

// direct-slot variant: the combined [G, H] grads land straight in the
// flat-grad views (no clones, no AccumulateGrad adds)
torch::Tensor layernorm_bwd_slots(torch::Tensor gy, torch::Tensor x,
                                  torch::Tensor w, torch::Tensor mean,
                                  torch::Tensor invstd, int64_t G,
                                  torch::Tensor gw_slot,
                                  torch::Tensor gb_slot) {
  check_bf16(gy, "gy");
  check_f32(gw_slot, "gw_slot");
  check_f32(gb_slot, "gb_slot");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto gc = gy.contiguous(), xc = x.contiguous();
  auto mc = mean.contiguous(), ic = invstd.contiguous();
  const int H = static_cast<int>(x.size(-1));
  const int64_t rows = x.numel() / H;
  auto gx = torch::empty_like(xc);
  auto gw_gb = torch::empty({G, 2 * H},
                            x.options().dtype(torch::kFloat32));
  auto partials = torch::empty({G * 64 * 2 * H},
                               x.options().dtype(torch::kFloat32));
  launch_layernorm_bwd(
      gc.data_ptr(), xc.data_ptr(), w.data_ptr<float>(),
      mc.data_ptr<float>(), ic.data_ptr<float>(), gx.data_ptr(),
      gw_gb.data_ptr<float>(), partials.data_ptr<float>(), rows, H,
      static_cast<int>(G), stream());
  // overwrite the interleaved combine with a direct-to-slot combine
  launch_ln_combine_slots(partials.data_ptr<float>(),
                          gw_slot.data_ptr<float>(),
                          gb_slot.data_ptr<float>(), H,
                          static_cast<int>(G), 64, stream());
  HIP_CHECK_LAST();
  return gx;
}

// ---- BatchNorm (channels-last) -------------------------------------------
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> batchnorm_fwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor b,
    torch::Tensor running_mean, torch::Tensor running_var,
    torch::Tensor residual, bool training, double momentum, double eps,
    bool relu, int64_t G) {
  check_bf16(x, "x");
  TORCH_CHECK(x.dim() == 4, "batchnorm expects NCHW tensor");
  const int C = static_cast<int>(x.size(1));
  TORCH_CHECK(C % 8 == 0, "native batchnorm needs C % 8 == 0, got ", C);
  TORCH_CHECK(x.size(0) % G == 0, "stacked batch must divide by G");
  TORCH_CHECK(w.numel() == G * C, "stacked BN weight must be [G*C]");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  const bool has_res = residual.numel() > 0;
  torch::Tensor rc;
  if (has_res) rc = residual.contiguous(torch::MemoryFormat::ChannelsLast);
  const int64_t R = x.numel() / C / G;  // rows per client
  auto y = torch::empty_like(xc);
  auto mean = torch::empty({G * C}, x.options().dtype(torch::kFloat32));
  auto invstd = torch::empty_like(mean);
  const int rgrid = norms_reduce_grid(R * C / 8);
  auto sums = torch::empty(
      {G * static_cast<int64_t>(rgrid + 1) * 2 * C}, mean.options());
  launch_batchnorm_fwd(
      xc.data_ptr(), has_res ? rc.data_ptr() : nullptr,
      w.data_ptr<float>(), b.data_ptr<float>(),
      running_mean.data_ptr<float>(), running_var.data_ptr<float>(),
      y.data_ptr(), mean.data_ptr<float>(), invstd.data_ptr<float>(),
      sums.data_ptr<float>(), R, C, static_cast<int>(G),
      static_cast<float>(momentum), static_cast<float>(eps), training,
      relu, stream());
  HIP_CHECK_LAST();
  return {y, mean, invstd};
}

// out_gw/out_gb: optional destinations (flat-grad views).  When given,
// the combine kernel writes the param grads STRAIGHT into them — the
// autograd Function then returns None for w/b and the per-step
// AccumulateGrad add kernels (40 of ~4.5us each per ResNet-18 step)
// never run.  Empty tensors => allocate and return plane views.
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
batchnorm_bwd(torch::Tensor gy, torch::Tensor y, torch::Tensor x,
              torch::Tensor w, torch::Tensor mean, torch::Tensor invstd,
              bool training, bool relu, int64_t G, torch::Tensor out_gw,
              torch::Tensor out_gb) {
  check_bf16(gy, "gy");
  const int C = static_cast<int>(x.size(1));
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  torch::Tensor yc;
  if (relu) yc = y.contiguous(torch::MemoryFormat::ChannelsLast);
  const int64_t R = x.numel() / C / G;
  auto gx = torch::empty_like(xc);
  torch::Tensor gym;  // masked gy == the residual grad
  if (relu) gym = torch::empty_like(gc);
  const int rgrid = norms_reduce_grid(R * C / 8);
  auto acc = torch::empty({G * static_cast<int64_t>(rgrid) * 2 * C},
                          x.options().dtype(torch::kFloat32));
  // plane layout [2][G*C]: row 0 = gb, row 1 = gw — contiguous views,
  // no clone kernels (see partial_combine_planes_kernel)
  torch::Tensor t_gw = out_gw, t_gb = out_gb;
  if (t_gw.numel() == 0) {
    auto gwgb = torch::empty({2, G * C},
                             x.options().dtype(torch::kFloat32));
    t_gb = gwgb.select(0, 0);
    t_gw = gwgb.select(0, 1);
  }
  TORCH_CHECK(t_gw.is_contiguous() && t_gb.is_contiguous(),
              "bn grad destinations must be contiguous");
  launch_batchnorm_bwd(gc.data_ptr(), relu ? yc.data_ptr() : nullptr,
                       xc.data_ptr(), w.data_ptr<float>(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gx.data_ptr(), relu ? gym.data_ptr() : nullptr,
                       acc.data_ptr<float>(), t_gb.data_ptr<float>(),
                       t_gw.data_ptr<float>(), R, C,
                       static_cast<int>(G), training, relu, stream());
  HIP_CHECK_LAST();
  return {gx, t_gw, t_gb, relu ? gym : gc};
}

// ---- GEMM family ----------------------------------------------------------
// all four operand layouts hit one kernel with transposed LDS staging —
// no host transpose passes.
torch::Tensor gemm2(torch::Tensor a, torch::Tensor b, bool a_km, bool b_km,
                    int64_t M, int64_t N, int64_t K,
                    const float* bias = nullptr) {
  check_bf16(a, "a");
  check_bf16(b, "b");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(a.device());
  auto ac = a.contiguous(), bc = b.contiguous();
  const int64_t batch = a.dim() == 3 ? a.size(0) : 1;
  // keep the caller's dimensionality: a 3-D batch-1 bmm must return 3-D
  // (autograd feeds the output shape back into the backward bmm calls)
  auto c = a.dim() == 3 ? torch::empty({batch, M, N}, a.options())
                        : torch::empty({M, N}, a.options());
  launch_gemm(ac.data_ptr(), bc.data_ptr(), c.data_ptr(), bias, batch, M,
              N, K, a_km, b_km, stream());
  HIP_CHECK_LAST();
  return c;
}

// linear forward: x @ w^T + bias with the bias fused into the epilogue
torch::Tensor linear_nt(torch::Tensor x, torch::Tensor w,
                        torch::Tensor bias) {
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1),
              "linear_nt shape mismatch");
  torch::Tensor bf;
  const float* bp = nullptr;
  if (bias.numel() > 0) {
    bf = bias.to(torch::kFloat32).contiguous();
    bp = bf.data_ptr<float>();
  }
  return gemm2(x, w, true, true, x.size(0), w.size(0), x.size(1), bp);
}

torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1),
              "gemm_nt shape mismatch");
  return gemm2(a, b, true, true, a.size(0), b.size(0), a.size(1));
}

// measured (BERT backward shapes): for 2-D flavors an explicit transpose
// pass + the vector-staged NT kernel beats the scatter-staged K-strided
// kernel by 15-30%; for small-batch bmm the native kernels win 2x.
torch::Tensor gemm_nn(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(0),
              "gemm_nn shape mismatch");
  auto bt = b.transpose(0, 1).contiguous();
  return gemm2(a, bt, true, true, a.size(0), b.size(1), a.size(1));
}

torch::Tensor gemm_tn(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(0) == b.size(0),
              "gemm_tn shape mismatch");
  auto at = a.transpose(0, 1).contiguous();
  auto bt = b.transpose(0, 1).contiguous();
  return gemm2(at, bt, true, true, a.size(1), b.size(1), a.size(0));
}

torch::Tensor bmm_nt(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.dim() == 3 && b.dim() == 3, "bmm expects 3-D");
  return gemm2(a, b, true, true, a.size(1), b.size(1), a.size(2));
}

torch::Tensor bmm_nn(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.dim() == 3 && b.dim() == 3, "bmm expects 3-D");
  return gemm2(a, b, true, false, a.size(1), b.size(2), a.size(2));
}

torch::Tensor bmm_tn(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.dim() == 3 && b.dim() == 3, "bmm expects 3-D");
  return gemm2(a, b, false, false, a.size(2), b.size(2), a.size(1));
}

// a @ b^T + bias, per-batch bias rows [batch, N] fused into the epilogue
// (stacked-client linear forward)
torch::Tensor bmm_nt_bias(torch::Tensor a, torch::Tensor b,
                          torch::Tensor bias) {
  TORCH_CHECK(a.dim() == 3 && b.dim() == 3, "bmm expects 3-D");
  check_bf16(a, "a");
  check_bf16(b, "b");
  check_f32(bias, "bias");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(a.device());
  auto ac = a.contiguous(), bc = b.contiguous(),
       bs = bias.contiguous();
  const int64_t batch = a.size(0), M = a.size(1), N = b.size(1),
                K = a.size(2);
  TORCH_CHECK(bs.numel() == batch * N, "bias must be [batch, N]");
  auto c = torch::empty({batch, M, N}, a.options());
  launch_gemm(ac.data_ptr(), bc.data_ptr(), c.data_ptr(),
              bs.data_ptr<float>(), batch, M, N, K, true, true, stream(),
              false, N);
  HIP_CHECK_LAST();
  return c;
}

// a^T @ b per batch with FP32 output written straight into ``out`` (the
// flat-grad slot): the stacked-linear weight gradient with no bf16
// round trip and no AccumulateGrad add.
void bmm_tn_f32(torch::Tensor a, torch::Tensor b, torch::Tensor out) {
  TORCH_CHECK(a.dim() == 3 && b.dim() == 3, "bmm expects 3-D");
  check_bf16(a, "a");
  check_bf16(b, "b");
  check_f32(out, "out");
  TORCH_CHECK(out.is_contiguous(), "out must be contiguous");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(a.device());
  auto ac = a.contiguous(), bc = b.contiguous();
  const int64_t batch = a.size(0), M = a.size(2), N = b.size(2),
                K = a.size(1);
  TORCH_CHECK(out.numel() == batch * M * N, "out shape mismatch");
  launch_gemm(ac.data_ptr(), bc.data_ptr(), out.data_ptr(), nullptr,
              batch, M, N, K, false, false, stream(), true, 0);
  HIP_CHECK_LAST();
}

// ---- conv family ----------------------------------------------------------
// weight 4-D [CO,CI,R,S] = single model; 5-D [G,CO,CI,R,S] = stacked
// clients (x batch is then G*B, client-major).
ConvShape make_shape(const torch::Tensor& x, const torch::Tensor& w,
                     int64_t stride, int64_t pad, int64_t G) {
  ConvShape cs;
  cs.N = static_cast<int>(x.size(0) / G);  // batch per client
  cs.C = static_cast<int>(x.size(1));
  cs.H = static_cast<int>(x.size(2));
  cs.W = static_cast<int>(x.size(3));
  const int wd = w.dim();
  cs.K = static_cast<int>(w.size(wd - 4));
  cs.R = static_cast<int>(w.size(wd - 2));
  cs.S = static_cast<int>(w.size(wd - 1));
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  cs.P = (cs.H + 2 * cs.pad - cs.R) / cs.stride + 1;
  cs.Q = (cs.W + 2 * cs.pad - cs.S) / cs.stride + 1;
  conv_shape_fill_magic(cs);
  return cs;
}

// Channel pad to 8 for C<8 inputs (the ResNet stem's C=3): the conv
// gathers only vectorize when C%8==0 — the scalar fallback made the stem
// the slowest wgrad of the whole model (measured 312us vs ~60 padded).
// Zero-padding x and w changes no output value; wgrad results for the pad
// channels are exact zeros and get sliced away.
static torch::Tensor pad_x_c8(const torch::Tensor& x) {
  auto y = torch::empty({x.size(0), 8, x.size(2), x.size(3)},
                        x.options().memory_format(
                            torch::MemoryFormat::ChannelsLast));
  y.narrow(1, x.size(1), 8 - x.size(1)).zero_();
  y.narrow(1, 0, x.size(1)).copy_(x);
  return y;
}

static torch::Tensor pad_w_c8(const torch::Tensor& w) {
  // K-major weight [*, CO, R, S, CI] -> CI padded to 8 (contiguous pad)
  auto sizes = w.sizes().vec();
  const int64_t ci = sizes.back();
  sizes.back() = 8;
  auto y = torch::empty(sizes, w.options());
  y.narrow(-1, ci, 8 - ci).zero_();
  y.narrow(-1, 0, ci).copy_(w);
  return y;
}

// shape helper for K-major weights ([*, CO, R, S, CI])
ConvShape make_shape_km(const torch::Tensor& x, const torch::Tensor& w,
                        int64_t stride, int64_t pad, int64_t G) {
  ConvShape cs;
  cs.N = static_cast<int>(x.size(0) / G);
  cs.C = static_cast<int>(x.size(1));
  cs.H = static_cast<int>(x.size(2));
  cs.W = static_cast<int>(x.size(3));
  const int wd = w.dim();
  cs.K = static_cast<int>(w.size(wd - 4));
  cs.R = static_cast<int>(w.size(wd - 3));
  cs.S = static_cast<int>(w.size(wd - 2));
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  cs.P = (cs.H + 2 * cs.pad - cs.R) / cs.stride + 1;
  cs.Q = (cs.W + 2 * cs.pad - cs.S) / cs.stride + 1;
  conv_shape_fill_magic(cs);
  return cs;
}

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                         int64_t pad) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  const int64_t G = w.dim() == 5 ? w.size(0) : 1;
  ConvShape cs = make_shape(x, w, stride, pad, G);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  // repack weight K-major: [(G)][CO][R][S][CI]
  auto wp = (w.dim() == 5 ? w.permute({0, 1, 3, 4, 2})
                          : w.permute({0, 2, 3, 1}))
                .contiguous();
  auto y = torch::empty({x.size(0), cs.K, cs.P, cs.Q},
                        x.options().memory_format(
                            torch::MemoryFormat::ChannelsLast));
  launch_conv_fwd(xc.data_ptr(), wp.data_ptr(), y.data_ptr(), cs,
                  static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  return y;
}

// K-major weight API (the model path: weights live as [*, CO, R, S, CI],
// so no permute pass per call)
torch::Tensor conv2d_fwd_km(torch::Tensor x, torch::Tensor w,
                            int64_t stride, int64_t pad) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  const int64_t G = w.dim() == 5 ? w.size(0) : 1;
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto wp = w.contiguous();
  if (x.size(1) < 8) {
    xc = pad_x_c8(xc);
    wp = pad_w_c8(wp);
  }
  ConvShape cs = make_shape_km(xc, wp, stride, pad, G);
  auto y = torch::empty({x.size(0), cs.K, cs.P, cs.Q},
                        x.options().memory_format(
                            torch::MemoryFormat::ChannelsLast));
  launch_conv_fwd(xc.data_ptr(), wp.data_ptr(), y.data_ptr(), cs,
                  static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor conv2d_dgrad_km(torch::Tensor gy, torch::Tensor w, int64_t H,
                              int64_t W, int64_t stride, int64_t pad) {
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(gy.device());
  const int64_t G = w.dim() == 5 ? w.size(0) : 1;
  const int wd = w.dim();
  ConvShape cs;
  cs.N = static_cast<int>(gy.size(0) / G);
  cs.K = static_cast<int>(gy.size(1));
  cs.P = static_cast<int>(gy.size(2));
  cs.Q = static_cast<int>(gy.size(3));
  cs.C = static_cast<int>(w.size(wd - 1));
  cs.R = static_cast<int>(w.size(wd - 3));
  cs.S = static_cast<int>(w.size(wd - 2));
  cs.H = static_cast<int>(H);
  cs.W = static_cast<int>(W);
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  conv_shape_fill_magic(cs);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  // wT K-major for dgrad: [(G)][CI][R][S][CO] (one permute; fwd has none)
  auto wt = (w.dim() == 5 ? w.permute({0, 4, 2, 3, 1})
                          : w.permute({3, 1, 2, 0}))
                .contiguous();
  auto gx = torch::empty({gy.size(0), cs.C, cs.H, cs.W},
                         gy.options().memory_format(
                             torch::MemoryFormat::ChannelsLast));
  launch_conv_dgrad(gc.data_ptr(), wt.data_ptr(), gx.data_ptr(), cs,
                    static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  return gx;
}

// K-major wgrad: returns [(G,) CO, R, S, CI] — the kernel's native
// output layout, no permute
torch::Tensor conv2d_wgrad_km(torch::Tensor gy, torch::Tensor x, int64_t R,
                              int64_t S, int64_t stride, int64_t pad,
                              int64_t G) {
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(gy.device());
  const int64_t ci = x.size(1);
  if (ci < 8) x = pad_x_c8(x.contiguous(torch::MemoryFormat::ChannelsLast));
  ConvShape cs;
  cs.N = static_cast<int>(x.size(0) / G);
  cs.C = static_cast<int>(x.size(1));
  cs.H = static_cast<int>(x.size(2));
  cs.W = static_cast<int>(x.size(3));
  cs.K = static_cast<int>(gy.size(1));
  cs.P = static_cast<int>(gy.size(2));
  cs.Q = static_cast<int>(gy.size(3));
  cs.R = static_cast<int>(R);
  cs.S = static_cast<int>(S);
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  conv_shape_fill_magic(cs);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  const int64_t rsc = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  const int splitk = conv_wgrad_splitk(cs, static_cast<int>(G));
  // splitk == 1 writes the final dtype straight from the epilogue —
  // don't reserve the (up to ~180 GB at G=4096) slab
  auto gwf = torch::empty(
      {splitk > 1 ? G * splitk : 0, cs.K, rsc},
      x.options().dtype(torch::kFloat32));
  auto gw = torch::empty({G, cs.K, cs.R, cs.S, cs.C}, x.options());
  launch_conv_wgrad(gc.data_ptr(), xc.data_ptr(), gwf.data_ptr<float>(),
                    gw.data_ptr(), cs, static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  if (ci < cs.C) gw = gw.narrow(-1, 0, ci).contiguous();
  return G == 1 ? gw.squeeze(0) : gw;
}

// fp32-grad variant for the shadow-weight path: grads are produced in the
// flat fp32 layout directly (no bf16 round trip on the weight gradient).
// out (optional, empty = allocate): the flat-grad view of the parameter —
// writing it here skips the autograd AccumulateGrad add per conv.
torch::Tensor conv2d_wgrad_km32(torch::Tensor gy, torch::Tensor x,
                                int64_t R, int64_t S, int64_t stride,
                                int64_t pad, int64_t G, torch::Tensor out) {
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(gy.device());
  const int64_t ci = x.size(1);
  if (ci < 8) x = pad_x_c8(x.contiguous(torch::MemoryFormat::ChannelsLast));
  ConvShape cs;
  cs.N = static_cast<int>(x.size(0) / G);
  cs.C = static_cast<int>(x.size(1));
  cs.H = static_cast<int>(x.size(2));
  cs.W = static_cast<int>(x.size(3));
  cs.K = static_cast<int>(gy.size(1));
  cs.P = static_cast<int>(gy.size(2));
  cs.Q = static_cast<int>(gy.size(3));
  cs.R = static_cast<int>(R);
  cs.S = static_cast<int>(S);
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  conv_shape_fill_magic(cs);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  const int64_t rsc = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  const int splitk = conv_wgrad_splitk(cs, static_cast<int>(G));
  // splitk == 1 writes the final dtype straight from the epilogue —
  // don't reserve the (up to ~180 GB at G=4096) slab
  auto gwf = torch::empty(
      {splitk > 1 ? G * splitk : 0, cs.K, rsc},
      x.options().dtype(torch::kFloat32));
  const bool direct = out.numel() > 0 && ci == cs.C;
  torch::Tensor gw;
  if (direct) {
    TORCH_CHECK(out.is_contiguous(), "wgrad destination must be contiguous");
    TORCH_CHECK(out.numel() == G * cs.K * rsc, "wgrad destination size");
    gw = out;
  } else {
    gw = torch::empty({G, cs.K, cs.R, cs.S, cs.C},
                      x.options().dtype(torch::kFloat32));
  }
  launch_conv_wgrad_out(gc.data_ptr(), xc.data_ptr(), gwf.data_ptr<float>(),
                        gw.data_ptr(), cs, static_cast<int>(G), true,
                        stream());
  HIP_CHECK_LAST();
  if (ci < cs.C) {
    gw = gw.narrow(-1, 0, ci);
    if (out.numel() > 0) {
      out.view({G, cs.K, cs.R, cs.S, ci}).copy_(gw);
      return out;
    }
    gw = gw.contiguous();
  }
  return G == 1 ? gw.squeeze(0) : gw;
}

torch::Tensor conv2d_dgrad(torch::Tensor gy, torch::Tensor w, int64_t H,
                           int64_t W, int64_t stride, int64_t pad) {
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(gy.device());
  const int64_t G = w.dim() == 5 ? w.size(0) : 1;
  const int wd = w.dim();
  ConvShape cs;
  cs.N = static_cast<int>(gy.size(0) / G);
  cs.K = static_cast<int>(gy.size(1));
  cs.P = static_cast<int>(gy.size(2));
  cs.Q = static_cast<int>(gy.size(3));
  cs.C = static_cast<int>(w.size(wd - 3));
  cs.R = static_cast<int>(w.size(wd - 2));
  cs.S = static_cast<int>(w.size(wd - 1));
  cs.H = static_cast<int>(H);
  cs.W = static_cast<int>(W);
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  conv_shape_fill_magic(cs);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  // wT K-major: [(G)][CI][R][S][CO]
  auto wt = (w.dim() == 5 ? w.permute({0, 2, 3, 4, 1})
                          : w.permute({1, 2, 3, 0}))
                .contiguous();
  auto gx = torch::empty({gy.size(0), cs.C, cs.H, cs.W},
                         gy.options().memory_format(
                             torch::MemoryFormat::ChannelsLast));
  launch_conv_dgrad(gc.data_ptr(), wt.data_ptr(), gx.data_ptr(), cs,
                    static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  return gx;
}

torch::Tensor conv2d_wgrad_fp32(torch::Tensor gy, torch::Tensor x,
                                int64_t R, int64_t S, int64_t stride,
                                int64_t pad, int64_t G) {
  // per-sample gradient path: returns the fp32 split-K sum directly
  // (splitk == 1 there), skipping the bf16 cast + re-float round trip
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(gy.device());
  ConvShape cs;
  cs.N = static_cast<int>(x.size(0) / G);
  cs.C = static_cast<int>(x.size(1));
  cs.H = static_cast<int>(x.size(2));
  cs.W = static_cast<int>(x.size(3));
  cs.K = static_cast<int>(gy.size(1));
  cs.P = static_cast<int>(gy.size(2));
  cs.Q = static_cast<int>(gy.size(3));
  cs.R = static_cast<int>(R);
  cs.S = static_cast<int>(S);
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  conv_shape_fill_magic(cs);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  // always splitk == 1 here: per-sample launches already provide
  // G = batch blocks of z-parallelism
  auto gwf = torch::empty({G, cs.K, cs.R, cs.S, cs.C},
                          x.options().dtype(torch::kFloat32));
  launch_conv_wgrad_nosum(gc.data_ptr(), xc.data_ptr(),
                          gwf.data_ptr<float>(), cs, static_cast<int>(G),
                          stream());
  HIP_CHECK_LAST();
  return gwf.permute({0, 1, 4, 2, 3}).contiguous();
}

// K-major per-sample wgrad: [B, CO, R, S, CI] fp32 — the kernel's native
// output, zero permute passes (matches K-major module weights)
torch::Tensor conv2d_wgrad_fp32_km(torch::Tensor gy, torch::Tensor x,
                                   int64_t R, int64_t S, int64_t stride,
                                   int64_t pad, int64_t G) {
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(gy.device());
  const int64_t ci = x.size(1);
  if (ci < 8) x = pad_x_c8(x.contiguous(torch::MemoryFormat::ChannelsLast));
  ConvShape cs;
  cs.N = static_cast<int>(x.size(0) / G);
  cs.C = static_cast<int>(x.size(1));
  cs.H = static_cast<int>(x.size(2));
  cs.W = static_cast<int>(x.size(3));
  cs.K = static_cast<int>(gy.size(1));
  cs.P = static_cast<int>(gy.size(2));
  cs.Q = static_cast<int>(gy.size(3));
  cs.R = static_cast<int>(R);
  cs.S = static_cast<int>(S);
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  conv_shape_fill_magic(cs);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto gwf = torch::empty({G, cs.K, cs.R, cs.S, cs.C},
                          x.options().dtype(torch::kFloat32));
  launch_conv_wgrad_nosum(gc.data_ptr(), xc.data_ptr(),
                          gwf.data_ptr<float>(), cs, static_cast<int>(G),
                          stream());
  HIP_CHECK_LAST();
  if (ci < cs.C) gwf = gwf.narrow(-1, 0, ci).contiguous();
  return gwf;
}

torch::Tensor conv2d_wgrad(torch::Tensor gy, torch::Tensor x, int64_t R,
                           int64_t S, int64_t stride, int64_t pad,
                           int64_t G) {
  check_bf16(gy, "gy");
  const c10::hip::HIPGuardMasqueradingAsCUDA guard(gy.device());
  ConvShape cs;
  cs.N = static_cast<int>(x.size(0) / G);
  cs.C = static_cast<int>(x.size(1));
  cs.H = static_cast<int>(x.size(2));
  cs.W = static_cast<int>(x.size(3));
  cs.K = static_cast<int>(gy.size(1));
  cs.P = static_cast<int>(gy.size(2));
  cs.Q = static_cast<int>(gy.size(3));
  cs.R = static_cast<int>(R);
  cs.S = static_cast<int>(S);
  cs.stride = static_cast<int>(stride);
  cs.pad = static_cast<int>(pad);
  conv_shape_fill_magic(cs);
  auto gc = gy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  const int64_t rsc = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  const int splitk = conv_wgrad_splitk(cs, static_cast<int>(G));
  // splitk == 1 writes the final dtype straight from the epilogue —
  // don't reserve the (up to ~180 GB at G=4096) slab
  auto gwf = torch::empty(
      {splitk > 1 ? G * splitk : 0, cs.K, rsc},
      x.options().dtype(torch::kFloat32));
  auto gw_rsci = torch::empty({G, cs.K, cs.R, cs.S, cs.C}, x.options());
  const char* env = getenv("COLEARN_WGRAD_VARIANT");
  const int variant = env ? atoi(env) : 0;
  if (variant)
    launch_conv_wgrad_variant(gc.data_ptr(), xc.data_ptr(),
                              gwf.data_ptr<float>(), gw_rsci.data_ptr(), cs,
                              static_cast<int>(G), variant, stream());
  else
    launch_conv_wgrad(gc.data_ptr(), xc.data_ptr(), gwf.data_ptr<float>(),
                      gw_rsci.data_ptr(), cs, static_cast<int>(G), stream());
  HIP_CHECK_LAST();
  auto out = gw_rsci.permute({0, 1, 4, 2, 3}).contiguous();
  return G == 1 ? out.squeeze(0) : out;
}

torch::Tensor tr_probe(int64_t scheme) {
  auto out = torch::empty({64, 4},
                          torch::device(torch::kCUDA).dtype(torch::kInt16));
  launch_tr_probe(out.data_ptr(), static_cast<int>(scheme), stream());
  HIP_CHECK_LAST();
  return out;
}

}  // namespace
}  // namespace colearn

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  using namespace colearn;
  m.def("fused_sgd", &fused_sgd);
  m.def("fused_adam", &fused_adam);
  m.def("sq_norm_per_row", &sq_norm_per_row);
  m.def("scale_rows_mean", &scale_rows_mean);
  m.def("philox_gaussian_add", &philox_gaussian_add);
  m.def("attn_fwd", &attn_fwd);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("colsum_f32", &colsum_f32);
  m.def("gn_grad_sample", &gn_grad_sample);
  m.def("sgn_fwd", &sgn_fwd);
  m.def("sgn_bwd", &sgn_bwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("head_fwd", &head_fwd);
  m.def("head_bwd", &head_bwd);
  m.def("shadow_sync", &shadow_sync);
  m.def("conv2d_dgrad_km_pre", &conv2d_dgrad_km_pre);
  m.def("topk_compress", &topk_compress);
  m.def("scatter_add_pairs", &scatter_add_pairs);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("layernorm_bwd_slots", &layernorm_bwd_slots);
  m.def("batchnorm_fwd", &batchnorm_fwd);
  m.def("batchnorm_bwd", &batchnorm_bwd);
  m.def("gemm_nt", &gemm_nt);
  m.def("linear_nt", &linear_nt);
  m.def("gemm_nn", &gemm_nn);
  m.def("gemm_tn", &gemm_tn);
  m.def("bmm_nt", &bmm_nt);
  m.def("bmm_nn", &bmm_nn);
  m.def("bmm_tn", &bmm_tn);
  m.def("bmm_nt_bias", &bmm_nt_bias);
  m.def("bmm_tn_f32", &bmm_tn_f32);
  m.def("conv2d_fwd", &conv2d_fwd);
  m.def("conv2d_fwd_km", &conv2d_fwd_km);
  m.def("conv2d_dgrad_km", &conv2d_dgrad_km);
  m.def("conv2d_wgrad_km", &conv2d_wgrad_km);
  m.def("conv2d_wgrad_km32", &conv2d_wgrad_km32);
  m.def("tr_probe", &tr_probe);
  m.def("conv2d_dgrad", &conv2d_dgrad);
  m.def("conv2d_wgrad", &conv2d_wgrad);
  m.def("conv2d_wgrad_fp32", &conv2d_wgrad_fp32);
  m.def("conv2d_wgrad_fp32_km", &conv2d_wgrad_fp32_km);
}
