// Small fused elementwise/reduction kernels for the BERT path (gfx950).
//
// torch's erf-GELU backward and bf16 column reductions measured far below
// bandwidth (0.4-0.8 TB/s) on the stacked BERT shapes
// (profiles/bert_stacked_fusedattn_prof.txt) — these are plain
// bandwidth-bound rewrites: 8-wide bf16 vectors, fp32 math, fixed-order
// accumulation (determinism contract).

#include "api.h"
#include "common.h"

namespace colearn {

// exact (erf) GELU, matching torch's default
DEV_INLINE float gelu_f(float x) {
  return 0.5f * x * (1.f + erff(x * 0.70710678118654752f));
}
DEV_INLINE float dgelu_f(float x) {
  const float cdf = 0.5f * (1.f + erff(x * 0.70710678118654752f));
  const float pdf = 0.3989422804014327f * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}

__global__ void gelu_fwd_kernel(const bf16* __restrict__ x,
                                bf16* __restrict__ y, int64_t n) {
  int64_t i = (static_cast<int64_t>(blockIdx.x) * blockDim.x +
               threadIdx.x) * 8;
  const int64_t stride = grid_stride() * 8;
  for (; i < n; i += stride) {
    if (i + 7 < n) {
      const short8 v = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(x) + i);
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o.x[j] = f2bits(gelu_f(bits2f(v.x[j])));
      *reinterpret_cast<short8*>(reinterpret_cast<short*>(y) + i) = o;
    } else {
      for (int64_t k = i; k < n; ++k) y[k] = f2b(gelu_f(b2f(x[k])));
    }
  }
}

__global__ void gelu_bwd_kernel(const bf16* __restrict__ gy,
                                const bf16* __restrict__ x,
                                bf16* __restrict__ gx, int64_t n) {
  int64_t i = (static_cast<int64_t>(blockIdx.x) * blockDim.x +
               threadIdx.x) * 8;
  const int64_t stride = grid_stride() * 8;
  for (; i < n; i += stride) {
    if (i + 7 < n) {
      const short8 g = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(gy) + i);
      const short8 v = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(x) + i);
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.x[j] = f2bits(bits2f(g.x[j]) * dgelu_f(bits2f(v.x[j])));
      *reinterpret_cast<short8*>(reinterpret_cast<short*>(gx) + i) = o;
    } else {
      for (int64_t k = i; k < n; ++k)
        gx[k] = f2b(b2f(gy[k]) * dgelu_f(b2f(x[k])));
    }
  }
}

// column sums of a bf16 [B, M, N] tensor over M -> fp32 [B, N]
// (stacked-linear bias gradients).  Deterministic: each output column is
// one thread accumulating rows in order; grid.y chunks rows into fixed
// slots combined in order by a second pass over the partials.
__global__ void colsum_partial_kernel(const bf16* __restrict__ x,
                                      float* __restrict__ partials,
                                      int64_t M, int64_t N, int chunks) {
  const int64_t b = blockIdx.z;
  const int64_t col = static_cast<int64_t>(blockIdx.x) * blockDim.x +
                      threadIdx.x;
  if (col >= N) return;
  const int64_t rows_per = (M + chunks - 1) / chunks;
  const int64_t r0 = static_cast<int64_t>(blockIdx.y) * rows_per;
  const int64_t r1 = min(r0 + rows_per, M);
  const short* xb = reinterpret_cast<const short*>(x) + b * M * N;
  float acc = 0.f;
  for (int64_t r = r0; r < r1; ++r) acc += bits2f(xb[r * N + col]);
  partials[(b * chunks + blockIdx.y) * N + col] = acc;
}

__global__ void colsum_combine_kernel(const float* __restrict__ partials,
                                      float* __restrict__ out, int64_t N,
                                      int chunks) {
  const int64_t b = blockIdx.y;
  const int64_t col = static_cast<int64_t>(blockIdx.x) * blockDim.x +
                      threadIdx.x;
  if (col >= N) return;
  float acc = 0.f;
  for (int c = 0; c < chunks; ++c)
    acc += partials[(b * chunks + c) * N + col];
  out[b * N + col] = acc;
}

// ---- launchers ------------------------------------------------------------
void launch_gelu_fwd(const void* x, void* y, int64_t n,
                     hipStream_t stream) {
  gelu_fwd_kernel<<<elemwise_grid((n + 7) / 8), 256, 0, stream>>>(
      static_cast<const bf16*>(x), static_cast<bf16*>(y), n);
}

void launch_gelu_bwd(const void* gy, const void* x, void* gx, int64_t n,
                     hipStream_t stream) {
  gelu_bwd_kernel<<<elemwise_grid((n + 7) / 8), 256, 0, stream>>>(
      static_cast<const bf16*>(gy), static_cast<const bf16*>(x),
      static_cast<bf16*>(gx), n);
}

void launch_colsum(const void* x, float* partials, float* out, int64_t B,
                   int64_t M, int64_t N, int chunks, hipStream_t stream) {
  dim3 g1(static_cast<unsigned>((N + 255) / 256), chunks,
          static_cast<unsigned>(B));
  colsum_partial_kernel<<<g1, 256, 0, stream>>>(
      static_cast<const bf16*>(x), partials, M, N, chunks);
  dim3 g2(static_cast<unsigned>((N + 255) / 256),
          static_cast<unsigned>(B));
  colsum_combine_kernel<<<g2, 256, 0, stream>>>(partials, out, N, chunks);
}

}  // namespace colearn
