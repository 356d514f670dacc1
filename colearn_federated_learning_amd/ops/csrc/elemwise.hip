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

// GroupNorm per-sample parameter gradients (stacked DP rule), fused:
//   gw[n,c] = sum_hw gy * xhat ,  gb[n,c] = sum_hw gy
// with xhat recomputed from per-(sample, group) stats — 3 bf16 passes
// over the activation instead of the ~10 fp32 passes of the torch
// composition (float casts + group_norm recompute + muls + reduces).
// x/gy are channels-last [N, C, H, W] bf16; one block per sample;
// deterministic fixed-order combines.
template <int BLOCK>
__global__ void gn_grad_sample_kernel(const bf16* __restrict__ x,
                                      const bf16* __restrict__ gy,
                                      const bf16* __restrict__ yout,
                                      float* __restrict__ gw,
                                      float* __restrict__ gb, int C,
                                      int HW, int groups, float eps,
                                      bool relu) {
  extern __shared__ float lds[];  // [2][max(spc*C, C)] partials
  const int n = blockIdx.x;
  const short* xs = reinterpret_cast<const short*>(x) +
                    static_cast<int64_t>(n) * HW * C;
  const short* gs = reinterpret_cast<const short*>(gy) +
                    static_cast<int64_t>(n) * HW * C;
  const short* yo = yout ? reinterpret_cast<const short*>(yout) +
                               static_cast<int64_t>(n) * HW * C
                         : nullptr;
  // C <= BLOCK: spc hw-splits per channel; C > BLOCK: one split, each
  // thread owns C/BLOCK channels.
  const int spc = C <= BLOCK ? BLOCK / C : 1;
  const int cpt = C <= BLOCK ? 1 : C / BLOCK;
  const int slots = spc * C;  // LDS floats per plane
#pragma unroll 1
  for (int t = 0; t < cpt; ++t) {
    const int c = (threadIdx.x % C) + t * BLOCK;
    const int sl = threadIdx.x / C;
    float s = 0.f, ss = 0.f;
    if (sl < spc && c < C) {
      for (int p = sl; p < HW; p += spc) {
        const float v = bits2f(xs[static_cast<int64_t>(p) * C + c]);
        s += v;
        ss += v * v;
      }
      lds[sl * C + c] = s;
      lds[slots + sl * C + c] = ss;
    }
  }
  __syncthreads();
  __shared__ float gmean[64], ginv[64];
  if (threadIdx.x < groups) {
    const int cpg = C / groups;
    float as = 0.f, assq = 0.f;
    for (int cc = threadIdx.x * cpg; cc < (threadIdx.x + 1) * cpg; ++cc)
      for (int k = 0; k < spc; ++k) {
        as += lds[k * C + cc];
        assq += lds[slots + k * C + cc];
      }
    const float cnt = static_cast<float>(cpg) * HW;
    const float mu = as / cnt;
    gmean[threadIdx.x] = mu;
    ginv[threadIdx.x] = rsqrtf(fmaxf(assq / cnt - mu * mu, 0.f) + eps);
  }
  __syncthreads();
#pragma unroll 1
  for (int t = 0; t < cpt; ++t) {
    const int c = (threadIdx.x % C) + t * BLOCK;
    const int sl = threadIdx.x / C;
    if (sl < spc && c < C) {
      const float mu = gmean[c / (C / groups)];
      const float inv = ginv[c / (C / groups)];
      float aw = 0.f, ab = 0.f;
      for (int p = sl; p < HW; p += spc) {
        const int64_t i = static_cast<int64_t>(p) * C + c;
        float g = bits2f(gs[i]);
        if (relu && bits2f(yo[i]) <= 0.f) g = 0.f;
        aw += g * (bits2f(xs[i]) - mu) * inv;
        ab += g;
      }
      if (spc == 1) {  // no cross-split combine needed
        gw[static_cast<int64_t>(n) * C + c] = aw;
        gb[static_cast<int64_t>(n) * C + c] = ab;
      } else {
        lds[sl * C + c] = aw;
        lds[slots + sl * C + c] = ab;
      }
    }
  }
  if (spc > 1) {
    __syncthreads();
    if (threadIdx.x < C) {
      float w = 0.f, b = 0.f;
      for (int k = 0; k < spc; ++k) {
        w += lds[k * C + threadIdx.x];
        b += lds[slots + k * C + threadIdx.x];
      }
      gw[static_cast<int64_t>(n) * C + threadIdx.x] = w;
      gb[static_cast<int64_t>(n) * C + threadIdx.x] = b;
    }
  }
}

// Stacked GroupNorm forward/backward (DP path): per-SAMPLE group stats,
// per-CLIENT affine (w/b [G, C], g = n / rows-per-client).  torch's
// composition ran the stats/normalize/affine and their backwards through
// fp32 internals (~40 ms/step at config 5).  Same block structure as
// gn_grad_sample; channels-last bf16; outputs mean/invstd [N, groups]
// fp32 for the backward.
template <int BLOCK>
__global__ void sgn_fwd_kernel(const bf16* __restrict__ x,
                               const float* __restrict__ w,
                               const float* __restrict__ b,
                               const bf16* __restrict__ res,
                               bf16* __restrict__ y,
                               float* __restrict__ mean_out,
                               float* __restrict__ invstd_out, int C,
                               int HW, int groups, int B_per_client,
                               float eps, bool relu) {
  extern __shared__ float lds[];
  const int n = blockIdx.x;
  const int g = n / B_per_client;  // client
  const short* xs = reinterpret_cast<const short*>(x) +
                    static_cast<int64_t>(n) * HW * C;
  short* ys = reinterpret_cast<short*>(y) +
              static_cast<int64_t>(n) * HW * C;
  const float* wg = w + static_cast<int64_t>(g) * C;
  const float* bg = b + static_cast<int64_t>(g) * C;
  const int spc = C <= BLOCK ? BLOCK / C : 1;
  const int cpt = C <= BLOCK ? 1 : C / BLOCK;
  const int slots = spc * C;
#pragma unroll 1
  for (int t = 0; t < cpt; ++t) {
    const int c = (threadIdx.x % C) + t * BLOCK;
    const int sl = threadIdx.x / C;
    if (sl < spc && c < C) {
      float s = 0.f, ss = 0.f;
      for (int p = sl; p < HW; p += spc) {
        const float v = bits2f(xs[static_cast<int64_t>(p) * C + c]);
        s += v;
        ss += v * v;
      }
      lds[sl * C + c] = s;
      lds[slots + sl * C + c] = ss;
    }
  }
  __syncthreads();
  __shared__ float gmean[64], ginv[64];
  if (threadIdx.x < groups) {
    const int cpg = C / groups;
    float as = 0.f, assq = 0.f;
    for (int cc = threadIdx.x * cpg; cc < (threadIdx.x + 1) * cpg; ++cc)
      for (int k = 0; k < spc; ++k) {
        as += lds[k * C + cc];
        assq += lds[slots + k * C + cc];
      }
    const float cnt = static_cast<float>(cpg) * HW;
    const float mu = as / cnt;
    gmean[threadIdx.x] = mu;
    const float inv = rsqrtf(fmaxf(assq / cnt - mu * mu, 0.f) + eps);
    ginv[threadIdx.x] = inv;
    mean_out[static_cast<int64_t>(n) * groups + threadIdx.x] = mu;
    invstd_out[static_cast<int64_t>(n) * groups + threadIdx.x] = inv;
  }
  __syncthreads();
#pragma unroll 1
  for (int t = 0; t < cpt; ++t) {
    const int c = (threadIdx.x % C) + t * BLOCK;
    const int sl = threadIdx.x / C;
    if (sl < spc && c < C) {
      const int grp = c / (C / groups);
      const float mu = gmean[grp], inv = ginv[grp];
      const float wc = wg[c], bc = bg[c];
      const short* rs =
          res ? reinterpret_cast<const short*>(res) +
                    static_cast<int64_t>(n) * HW * C
              : nullptr;
      for (int p = sl; p < HW; p += spc) {
        const int64_t i = static_cast<int64_t>(p) * C + c;
        float v = (bits2f(xs[i]) - mu) * inv * wc + bc;
        if (rs) v += bits2f(rs[i]);
        if (relu) v = fmaxf(v, 0.f);
        ys[i] = f2bits(v);
      }
    }
  }
}

// backward: gx = invstd * ( du - mean_grp(du) - xhat * mean_grp(du*xhat) )
// with du = gy * w_c per channel; means over the (C/groups x HW) group.
template <int BLOCK>
__global__ void sgn_bwd_kernel(const bf16* __restrict__ gy,
                               const bf16* __restrict__ x,
                               const bf16* __restrict__ yout,
                               const float* __restrict__ w,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               bf16* __restrict__ gx,
                               bf16* __restrict__ gres, int C, int HW,
                               int groups, int B_per_client, bool relu) {
  extern __shared__ float lds[];
  const int n = blockIdx.x;
  const int g = n / B_per_client;
  const short* gs = reinterpret_cast<const short*>(gy) +
                    static_cast<int64_t>(n) * HW * C;
  const short* xs = reinterpret_cast<const short*>(x) +
                    static_cast<int64_t>(n) * HW * C;
  short* os = reinterpret_cast<short*>(gx) +
              static_cast<int64_t>(n) * HW * C;
  const float* wg = w + static_cast<int64_t>(g) * C;
  const int spc = C <= BLOCK ? BLOCK / C : 1;
  const int cpt = C <= BLOCK ? 1 : C / BLOCK;
  const int slots = spc * C;
  __shared__ float gmu[64], ginv[64], g1[64], g2[64];
  if (threadIdx.x < groups) {
    gmu[threadIdx.x] = mean[static_cast<int64_t>(n) * groups + threadIdx.x];
    ginv[threadIdx.x] =
        invstd[static_cast<int64_t>(n) * groups + threadIdx.x];
  }
  __syncthreads();
#pragma unroll 1
  for (int t = 0; t < cpt; ++t) {
    const int c = (threadIdx.x % C) + t * BLOCK;
    const int sl = threadIdx.x / C;
    if (sl < spc && c < C) {
      const int grp = c / (C / groups);
      const float mu = gmu[grp], inv = ginv[grp], wc = wg[c];
      const short* yo = reinterpret_cast<const short*>(yout) +
                        static_cast<int64_t>(n) * HW * C;
      short* gr = gres ? reinterpret_cast<short*>(gres) +
                             static_cast<int64_t>(n) * HW * C
                       : nullptr;
      float s1 = 0.f, s2 = 0.f;  // sum(du), sum(du * xhat)
      for (int p = sl; p < HW; p += spc) {
        const int64_t i = static_cast<int64_t>(p) * C + c;
        float g = bits2f(gs[i]);
        if (relu && bits2f(yo[i]) <= 0.f) g = 0.f;
        if (gr) gr[i] = f2bits(g);
        const float du = g * wc;
        s1 += du;
        s2 += du * (bits2f(xs[i]) - mu) * inv;
      }
      lds[sl * C + c] = s1;
      lds[slots + sl * C + c] = s2;
    }
  }
  __syncthreads();
  if (threadIdx.x < groups) {
    const int cpg = C / groups;
    float a1 = 0.f, a2 = 0.f;
    for (int cc = threadIdx.x * cpg; cc < (threadIdx.x + 1) * cpg; ++cc)
      for (int k = 0; k < spc; ++k) {
        a1 += lds[k * C + cc];
        a2 += lds[slots + k * C + cc];
      }
    const float cnt = static_cast<float>(cpg) * HW;
    g1[threadIdx.x] = a1 / cnt;
    g2[threadIdx.x] = a2 / cnt;
  }
  __syncthreads();
#pragma unroll 1
  for (int t = 0; t < cpt; ++t) {
    const int c = (threadIdx.x % C) + t * BLOCK;
    const int sl = threadIdx.x / C;
    if (sl < spc && c < C) {
      const int grp = c / (C / groups);
      const float mu = gmu[grp], inv = ginv[grp], wc = wg[c];
      const float m1 = g1[grp], m2 = g2[grp];
      const short* yo = reinterpret_cast<const short*>(yout) +
                        static_cast<int64_t>(n) * HW * C;
      for (int p = sl; p < HW; p += spc) {
        const int64_t i = static_cast<int64_t>(p) * C + c;
        float g = bits2f(gs[i]);
        if (relu && bits2f(yo[i]) <= 0.f) g = 0.f;
        const float du = g * wc;
        const float xhat = (bits2f(xs[i]) - mu) * inv;
        os[i] = f2bits(inv * (du - m1 - xhat * m2));
      }
    }
  }
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

void launch_sgn_fwd(const void* x, const float* w, const float* b,
                    const void* res, void* y, float* mean, float* invstd,
                    int N, int C, int HW, int groups, int B_per_client,
                    float eps, bool relu, hipStream_t stream) {
  const int spc = C <= 256 ? 256 / C : 1;
  const size_t lds = 2 * static_cast<size_t>(spc * C > C ? spc * C : C) *
                     sizeof(float);
  sgn_fwd_kernel<256><<<N, 256, lds, stream>>>(
      static_cast<const bf16*>(x), w, b, static_cast<const bf16*>(res),
      static_cast<bf16*>(y), mean, invstd, C, HW, groups, B_per_client,
      eps, relu);
}

void launch_sgn_bwd(const void* gy, const void* x, const void* yout,
                    const float* w, const float* mean,
                    const float* invstd, void* gx, void* gres, int N,
                    int C, int HW, int groups, int B_per_client,
                    bool relu, hipStream_t stream) {
  const int spc = C <= 256 ? 256 / C : 1;
  const size_t lds = 2 * static_cast<size_t>(spc * C > C ? spc * C : C) *
                     sizeof(float);
  sgn_bwd_kernel<256><<<N, 256, lds, stream>>>(
      static_cast<const bf16*>(gy), static_cast<const bf16*>(x),
      static_cast<const bf16*>(yout), w, mean, invstd,
      static_cast<bf16*>(gx), static_cast<bf16*>(gres), C, HW, groups,
      B_per_client, relu);
}

void launch_gn_grad_sample(const void* x, const void* gy,
                           const void* yout, float* gw, float* gb, int N,
                           int C, int HW, int groups, float eps,
                           bool relu, hipStream_t stream) {
  const int spc = C <= 256 ? 256 / C : 1;
  const size_t lds = 2 * static_cast<size_t>(spc * C > C ? spc * C : C) *
                     sizeof(float);
  gn_grad_sample_kernel<256><<<N, 256, lds, stream>>>(
      static_cast<const bf16*>(x), static_cast<const bf16*>(gy),
      static_cast<const bf16*>(yout), gw, gb, C, HW, groups, eps, relu);
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
