// Fused stacked classifier head: avgpool + per-client FC + softmax-CE
// (forward AND backward), gfx950.
//
// Motivation: the un-fused head was ~15
// sub-5 µs launches per captured step — adaptive_avg_pool2d fwd/bwd,
// batched GEMM fwd + 2 bwd GEMMs, float casts, softmax, nll, view/mean
// chains — part of a measured ~20% small-kernel tail
// (profiles/resnet18_fedavg_kernel_stats.md).  Here the whole head is 4
// launches: fwd (pool+logits+CE probs), per-client loss reduce, bwd
// (dlogits+dx), wgrad (dW/db written straight into the flat-grad slots).
//
// Layouts: x / dx are channels-last bf16 [N, C, H, W] (the conv kernels'
// native activation layout), N = G*B client-major.  Weights [G, K, C]
// fp32 (the head GEMV reads master weights directly — more accurate than
// the bf16-shadow matmul it replaces, and tiny: K*C floats per client).
// All reductions are fixed-order => bitwise run-to-run deterministic.

#include "api.h"
#include "common.h"

namespace colearn {

// one workgroup per sample: pool C channels over HW, per-client GEMV to
// K logits, numerically-stable softmax CE; saves pooled + probs for bwd.
template <int BLOCK>
__global__ void head_fwd_kernel(const bf16* __restrict__ x,
                                const float* __restrict__ w,
                                const float* __restrict__ b,
                                const int64_t* __restrict__ y,
                                float* __restrict__ pooled,
                                float* __restrict__ probs,
                                float* __restrict__ loss_n,
                                int C, int HW, int K, int B) {
  extern __shared__ float lds[];           // [C] pooled + [BLOCK] scratch
  float* lds_pool = lds;
  float* scratch = lds + C;
  const int n = blockIdx.x;
  const int g = n / B;
  const float inv_hw = 1.f / HW;
  const bf16* xn = x + static_cast<int64_t>(n) * HW * C;

  for (int c = threadIdx.x; c < C; c += BLOCK) {
    float acc = 0.f;
    for (int p = 0; p < HW; ++p) acc += b2f(xn[p * C + c]);
    acc *= inv_hw;
    lds_pool[c] = acc;
    pooled[static_cast<int64_t>(n) * C + c] = acc;
  }
  __syncthreads();

  // logits: each thread accumulates K partials over its C slice
  constexpr int KMAX = 32;
  float part[KMAX];
  for (int k = 0; k < K; ++k) part[k] = 0.f;
  const float* wg = w + static_cast<int64_t>(g) * K * C;
  for (int c = threadIdx.x; c < C; c += BLOCK) {
    const float pv = lds_pool[c];
    for (int k = 0; k < K; ++k) part[k] += pv * wg[k * C + c];
  }
  // block tree-reduce each k (deterministic)
  for (int k = 0; k < K; ++k) {
    scratch[threadIdx.x] = part[k];
    __syncthreads();
    for (int s = BLOCK / 2; s > 0; s >>= 1) {
      if (threadIdx.x < s)
        scratch[threadIdx.x] += scratch[threadIdx.x + s];
      __syncthreads();
    }
    if (threadIdx.x == 0) lds_pool[k] = scratch[0];  // reuse pool LDS head
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    float logit[KMAX];
    float mx = -1e30f;
    for (int k = 0; k < K; ++k) {
      logit[k] = lds_pool[k] + b[g * K + k];
      mx = fmaxf(mx, logit[k]);
    }
    float denom = 0.f;
    for (int k = 0; k < K; ++k) denom += __expf(logit[k] - mx);
    const float lse = __logf(denom) + mx;
    const int target = static_cast<int>(y[n]);
    loss_n[n] = lse - logit[target];
    for (int k = 0; k < K; ++k)
      probs[static_cast<int64_t>(n) * K + k] = __expf(logit[k] - lse);
  }
}

// per-client mean loss, fixed-order tree (deterministic)
template <int BLOCK>
__global__ void head_loss_reduce_kernel(const float* __restrict__ loss_n,
                                        float* __restrict__ loss_vec,
                                        int B) {
  __shared__ float scratch[BLOCK];
  const int g = blockIdx.x;
  float acc = 0.f;
  for (int i = threadIdx.x; i < B; i += BLOCK) acc += loss_n[g * B + i];
  scratch[threadIdx.x] = acc;
  __syncthreads();
  for (int s = BLOCK / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) scratch[threadIdx.x] += scratch[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) loss_vec[g] = scratch[0] / B;
}

// bwd: dlogits = gloss[g] * (probs - onehot) / B ; dx = W^T dlogits / HW
// (broadcast over HW, bf16 channels-last).  One workgroup per sample.
template <int BLOCK>
__global__ void head_bwd_dx_kernel(const float* __restrict__ probs,
                                   const float* __restrict__ w,
                                   const int64_t* __restrict__ y,
                                   const float* __restrict__ gloss,
                                   float* __restrict__ dlogits,
                                   bf16* __restrict__ dx,
                                   int C, int HW, int K, int B) {
  constexpr int KMAX = 32;
  __shared__ float dl[KMAX];
  const int n = blockIdx.x;
  const int g = n / B;
  if (threadIdx.x < K) {
    const int k = threadIdx.x;
    const float scale = gloss[g] / B;
    float v = probs[static_cast<int64_t>(n) * K + k];
    if (k == static_cast<int>(y[n])) v -= 1.f;
    v *= scale;
    dl[k] = v;
    dlogits[static_cast<int64_t>(n) * K + k] = v;
  }
  __syncthreads();
  const float* wg = w + static_cast<int64_t>(g) * K * C;
  bf16* dxn = dx + static_cast<int64_t>(n) * HW * C;
  const float inv_hw = 1.f / HW;
  for (int c = threadIdx.x; c < C; c += BLOCK) {
    float acc = 0.f;
    for (int k = 0; k < K; ++k) acc += dl[k] * wg[k * C + c];
    const bf16 v = f2b(acc * inv_hw);
    for (int p = 0; p < HW; ++p) dxn[p * C + c] = v;
  }
}

// wgrad: dW[g,k,c] = sum_b dlogits[n,k] * pooled[n,c] (fixed b order),
// db[g,k] = sum_b dlogits[n,k].  Written STRAIGHT into the flat-grad
// slots (overwrite: the head params have no other grad contributors).
template <int BLOCK>
__global__ void head_wgrad_kernel(const float* __restrict__ dlogits,
                                  const float* __restrict__ pooled,
                                  float* __restrict__ gw,
                                  float* __restrict__ gb,
                                  int C, int K, int B) {
  const int g = blockIdx.x;
  const int k = blockIdx.y;
  float* out = gw + (static_cast<int64_t>(g) * K + k) * C;
  for (int c = threadIdx.x; c < C; c += BLOCK) {
    float acc = 0.f;
    for (int bi = 0; bi < B; ++bi) {
      const int64_t n = static_cast<int64_t>(g) * B + bi;
      acc += dlogits[n * K + k] * pooled[n * C + c];
    }
    out[c] = acc;
  }
  if (threadIdx.x == 0) {
    float acc = 0.f;
    for (int bi = 0; bi < B; ++bi)
      acc += dlogits[(static_cast<int64_t>(g) * B + bi) * K + k];
    gb[g * K + k] = acc;
  }
}

// ---- launchers ------------------------------------------------------------
void launch_head_fwd(const void* x, const float* w, const float* b,
                     const int64_t* y, float* pooled, float* probs,
                     float* loss_n, float* loss_vec, int N, int C, int HW,
                     int K, int B, int G, hipStream_t stream) {
  const size_t lds = (C + 256) * sizeof(float);
  head_fwd_kernel<256><<<N, 256, lds, stream>>>(
      static_cast<const bf16*>(x), w, b, y, pooled, probs, loss_n, C, HW,
      K, B);
  head_loss_reduce_kernel<256><<<G, 256, 0, stream>>>(loss_n, loss_vec, B);
}

void launch_head_bwd(const float* probs, const float* pooled,
                     const float* w, const int64_t* y, const float* gloss,
                     float* dlogits, void* dx, float* gw, float* gb,
                     int N, int C, int HW, int K, int B, int G,
                     hipStream_t stream) {
  head_bwd_dx_kernel<256><<<N, 256, 0, stream>>>(
      probs, w, y, gloss, dlogits, static_cast<bf16*>(dx), C, HW, K, B);
  dim3 grid(G, K);
  head_wgrad_kernel<256><<<grid, 256, 0, stream>>>(dlogits, pooled, gw,
                                                   gb, C, K, B);
}

}  // namespace colearn
