// Host-visible launcher API shared by bindings.cpp (compiled with g++) and
// the .hip translation units (compiled with hipcc).  bf16 pointers cross
// this boundary as void*.
#pragma once

#include <hip/hip_runtime_api.h>

#include <cstdint>

namespace colearn {

// u32 magic division (round-up variant): n/d == umulhi(n, m) >> sh for
// n < 2^31.  Division by runtime C/S/Q/... inside the gather loops costs
// ~20-40 cycles each; these make it ~3.
struct U32Div {
  unsigned m;
  int sh;
};

inline U32Div make_u32div(unsigned d) {
  // round-up variant with sh = floor(log2 d): m = ceil(2^(32+sh)/d) then
  // fits in 32 bits for non-powers-of-two (2^sh < d); powers of two use a
  // plain shift (m = 0 sentinel).  Exact for n < 2^31.
  U32Div r;
  if ((d & (d - 1)) == 0) {
    int sh = 0;
    while ((1u << sh) < d) ++sh;
    r.sh = sh;
    r.m = 0;
    return r;
  }
  int sh = 0;
  while ((2u << sh) < d) ++sh;  // sh = floor(log2 d)
  r.sh = sh;
  r.m = static_cast<unsigned>(
      ((static_cast<unsigned long long>(1) << (32 + sh)) + d - 1) / d);
  return r;
}

struct ConvShape {
  int N, H, W, C;  // input
  int K, R, S;     // out channels, filter
  int P, Q;        // output spatial
  int stride, pad;
  // precomputed divisors for the gather decompositions
  U32Div dC;    // k % C, k / C            (fwd: k -> ci)
  U32Div dSC;   // k / (S*C)               (fwd: k -> r)
  U32Div dK;    // k % K, k / K            (dgrad: k -> co)
  U32Div dSK;   // k / (S*K)               (dgrad: k -> r)
  U32Div dQ;    // m % Q                   (npq -> q)
  U32Div dPQ;   // m / (P*Q)               (npq -> n)
};

inline void conv_shape_fill_magic(ConvShape& cs) {
  cs.dC = make_u32div(cs.C);
  cs.dSC = make_u32div(cs.S * cs.C);
  cs.dK = make_u32div(cs.K);
  cs.dSK = make_u32div(cs.S * cs.K);
  cs.dQ = make_u32div(cs.Q);
  cs.dPQ = make_u32div(cs.P * cs.Q);
}

void launch_fused_sgd(float* w, const float* g, float* m, const float* ref,
                      int64_t n, int64_t ref_period, float lr,
                      float momentum, float wd, float mu, bool has_mom,
                      bool has_prox, hipStream_t stream);
void launch_fused_adam(float* w, const float* g, float* m, float* v,
                       const float* ref, const float* bc, int64_t n,
                       int64_t ref_period, float lr, float b1, float b2,
                       float eps, float wd, float mu, float bc1, float bc2,
                       bool has_prox, hipStream_t stream);
void launch_sq_norm_per_row(const void* g, bool bf16_in, float* sq,
                            float* partials, int64_t rows, int64_t cols,
                            int chunks, hipStream_t stream);
void launch_scale_rows_mean(const void* g, bool bf16_in,
                            const float* scale, float* out, int64_t rows,
                            int64_t cols, int groups, hipStream_t stream);
void launch_topk(const float* v, int64_t n, int64_t k, int* idx, float* val,
                 unsigned* workspace /* >= 1036 u32 */, int* cand,
                 int cand_cap, hipStream_t stream);
void launch_shadow_sync(const float* w, void* shadow, int64_t n,
                        const int64_t* segA, const int* segB, int nseg,
                        void* shadow_t, int64_t transpose_blocks,
                        hipStream_t stream);
void launch_head_fwd(const void* x, const float* w, const float* b,
                     const int64_t* y, float* pooled, float* probs,
                     float* loss_n, float* loss_vec, int N, int C, int HW,
                     int K, int B, int G, hipStream_t stream);
void launch_head_bwd(const float* probs, const float* pooled,
                     const float* w, const int64_t* y, const float* gloss,
                     float* dlogits, void* dx, float* gw, float* gb,
                     int N, int C, int HW, int K, int B, int G,
                     hipStream_t stream);
void launch_gelu_fwd(const void* x, void* y, int64_t n,
                     hipStream_t stream);
void launch_gelu_bwd(const void* gy, const void* x, void* gx, int64_t n,
                     hipStream_t stream);
void launch_sgn_fwd(const void* x, const float* w, const float* b,
                    const void* res, void* y, float* mean, float* invstd,
                    int N, int C, int HW, int groups, int B_per_client,
                    float eps, bool relu, hipStream_t stream);
void launch_sgn_bwd(const void* gy, const void* x, const void* yout,
                    const float* w, const float* mean,
                    const float* invstd, void* gx, void* gres, int N,
                    int C, int HW, int groups, int B_per_client,
                    bool relu, hipStream_t stream);
void launch_gn_grad_sample(const void* x, const void* gy,
                           const void* yout, float* gw, float* gb, int N,
                           int C, int HW, int groups, float eps,
                           bool relu, hipStream_t stream);
void launch_colsum(const void* x, float* partials, float* out, int64_t B,
                   int64_t M, int64_t N, int chunks, hipStream_t stream);
void launch_attn_fwd(const void* qkv, void* ctx, void* Psave,
                     unsigned long long* mask, int N, int S, int H,
                     float scale, float p, unsigned long long seed,
                     hipStream_t stream);
void launch_attn_bwd(const void* dctx, const void* qkv, const void* Psave,
                     const unsigned long long* mask, void* dS, void* dqkv,
                     int N, int S, int H, float scale, float p,
                     unsigned long long seed, hipStream_t stream);
void launch_philox_gaussian_add(float* out, int64_t n,
                                unsigned long long seed, float std,
                                hipStream_t stream);
void launch_scatter_add_pairs(float* out, const int* idx, const float* val,
                              int64_t k, hipStream_t stream);
void launch_layernorm_fwd(const void* x, const float* w, const float* b,
                          void* y, float* mean, float* invstd, int64_t rows,
                          int H, float eps, int G, hipStream_t stream);
int norms_reduce_grid(int64_t n8);
void launch_ln_combine_slots(const float* partials, float* gw, float* gb,
                             int H, int G, int nchunk,
                             hipStream_t stream);
void launch_layernorm_bwd(const void* gy, const void* x, const float* w,
                          const float* mean, const float* invstd, void* gx,
                          float* gw_gb, float* partials, int64_t rows, int H,
                          int G, hipStream_t stream);
void launch_batchnorm_fwd(const void* x, const void* res, const float* w,
                          const float* b, float* running_mean,
                          float* running_var, void* y, float* mean,
                          float* invstd, float* sums, int64_t R, int C,
                          int G, float momentum, float eps, bool training,
                          bool relu, hipStream_t stream);
void launch_batchnorm_bwd(const void* gy_, const void* y_, const void* x_,
                          const float* w, const float* mean,
                          const float* invstd, void* gx_, void* gym_,
                          float* acc, float* gwgb_b, float* gwgb_w,
                          int64_t R, int C, int G,
                          bool training, bool relu, hipStream_t stream);
void launch_gemm_nt(const void* A, const void* B, void* C, int64_t batch,
                    int64_t M, int64_t N, int64_t K, hipStream_t stream);
void launch_gemm(const void* A, const void* B, void* C, const float* bias,
                 int64_t batch, int64_t M, int64_t N, int64_t K, bool a_km,
                 bool b_km, hipStream_t stream, bool f32_out = false,
                 int64_t sBias = 0);
void launch_conv_fwd(const void* x, const void* w, void* y, ConvShape cs,
                     int G, hipStream_t stream);
void launch_conv_dgrad(const void* gy, const void* wt, void* gx,
                       ConvShape cs, int G, hipStream_t stream);
int conv_wgrad_splitk(ConvShape cs, int G);
void launch_conv_wgrad_nosum(const void* gy, const void* x, float* gwf,
                             ConvShape cs, int G, hipStream_t stream);
void launch_tr_probe(void* out, int scheme, hipStream_t stream);
void launch_conv_wgrad_out(const void* gy, const void* x, float* gwf,
                           void* gw, ConvShape cs, int G, bool f32_out,
                           hipStream_t stream);
void launch_conv_wgrad(const void* gy, const void* x, float* gwf, void* gw,
                       ConvShape cs, int G, hipStream_t stream);
void launch_conv_wgrad_variant(const void* gy, const void* x, float* gwf,
                               void* gw, ConvShape cs, int G, int variant,
                               hipStream_t stream);

}  // namespace colearn
