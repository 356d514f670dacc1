// DP-FedAvg + top-k sparsification kernels.
//
// sq_norm_per_row / scale_rows_mean: the fused per-sample clip pipeline —
// per-sample grad norms (one reduction sweep over [B, D]), then
// clip-scale + batch-mean in a second sweep.  Both are pure-bandwidth
// float4 kernels.
//
// topk_compress: 16-bit radix-select over the |v| bit pattern (two 256-bin
// histogram rounds + ticketed compaction) — no sort, O(3) sweeps of the
// data, exact k output.  scatter_add_pairs: the decompression side.

#include "api.h"
#include "common.h"

namespace colearn {

// ---- per-sample squared norms: sq[b] += sum_d g[b,d]^2 --------------------
// deterministic AND parallel: per-(row, chunk) partials in fixed slots,
// combined per row in chunk order.
template <typename T>
DEV_INLINE float ld_f(const T* p) {
  if constexpr (sizeof(T) == 2)
    return b2f(*reinterpret_cast<const bf16*>(p));
  else
    return *reinterpret_cast<const float*>(p);
}

template <typename T>
__global__ void sq_norm_partial_kernel(const T* __restrict__ g,
                                       float* __restrict__ partials,
                                       int64_t rows, int64_t cols,
                                       int chunks) {
  __shared__ float scratch[4];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const T* grow = g + row * cols;
  float acc = 0.f;
  int64_t i = (static_cast<int64_t>(blockIdx.y) * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = static_cast<int64_t>(gridDim.y) * blockDim.x * 4;
  for (; i + 3 < cols; i += stride) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float v = ld_f(grow + i + j);
      acc += v * v;
    }
  }
  if (blockIdx.y == 0 && threadIdx.x < (cols & 3)) {
    const float v = ld_f(grow + (cols & ~int64_t(3)) + threadIdx.x);
    acc += v * v;
  }
  acc = block_allreduce_sum<256>(acc, scratch);
  if (threadIdx.x == 0)
    partials[static_cast<int64_t>(row) * chunks + blockIdx.y] = acc;
}

__global__ void sq_norm_combine_kernel(const float* __restrict__ partials,
                                       float* __restrict__ sq, int64_t rows,
                                       int chunks) {
  const int64_t row = static_cast<int64_t>(blockIdx.x) * blockDim.x +
                      threadIdx.x;
  if (row >= rows) return;
  float acc = 0.f;
  for (int c = 0; c < chunks; ++c)
    acc += partials[row * chunks + c];
  sq[row] += acc;
}

// ---- out[d] = (1/rows) * sum_b scale[b] * g[b,d] --------------------------
// rows are split into gridDim.z equal GROUPS (stacked DP clients): each
// z writes its group's mean into out[z * cols ..]
template <typename T>
__global__ void scale_rows_mean_kernel(const T* __restrict__ g,
                                       const float* __restrict__ scale,
                                       float* __restrict__ out, int64_t rows,
                                       int64_t cols) {
  const int64_t rpg = rows / gridDim.z;
  g += static_cast<int64_t>(blockIdx.z) * rpg * cols;
  scale += static_cast<int64_t>(blockIdx.z) * rpg;
  out += static_cast<int64_t>(blockIdx.z) * cols;
  rows = rpg;
  int64_t d = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  const float inv = 1.f / static_cast<float>(rows);
  const int64_t stride = static_cast<int64_t>(blockDim.x) * gridDim.x * 4;
  for (; d + 3 < cols; d += stride) {
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    for (int64_t b = 0; b < rows; ++b) {
      const float s = scale[b];
      const T* p = g + b * cols + d;
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[j] += s * ld_f(p + j);
    }
    *reinterpret_cast<float4*>(out + d) = make_float4(
        acc[0] * inv, acc[1] * inv, acc[2] * inv, acc[3] * inv);
  }
  if (blockIdx.x == 0 && threadIdx.x < (cols & 3)) {
    const int64_t j = (cols & ~int64_t(3)) + threadIdx.x;
    float acc = 0.f;
    for (int64_t b = 0; b < rows; ++b)
      acc += scale[b] * ld_f(g + b * cols + j);
    out[j] = acc * inv;
  }
}

// ---- top-k radix select ---------------------------------------------------
// key(v) = top 16 bits of |v|'s f32 bit pattern (monotone in |v|).

DEV_INLINE unsigned absbits(float v) {
  union {
    float f;
    unsigned u;
  } c;
  c.f = v;
  return c.u & 0x7FFFFFFFu;
}

__global__ void topk_hist1_kernel(const float* __restrict__ v, int64_t n,
                                  unsigned* __restrict__ hist /*[256]*/) {
  __shared__ unsigned lh[256];
  if (threadIdx.x < 256) lh[threadIdx.x] = 0;
  __syncthreads();
  int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (; i < n; i += grid_stride())
    atomicAdd(&lh[absbits(v[i]) >> 24], 1u);
  __syncthreads();
  if (threadIdx.x < 256 && lh[threadIdx.x])
    atomicAdd(hist + threadIdx.x, lh[threadIdx.x]);
}

// pick the bin where the k-th largest falls; single block of 256.
// out: sel[0]=bin, sel[1]=count strictly above bin, sel[2]=k remaining in bin
__global__ void topk_select_kernel(const unsigned* __restrict__ hist,
                                   int64_t k, unsigned* __restrict__ sel) {
  __shared__ unsigned h[256];
  h[threadIdx.x] = hist[threadIdx.x];
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned above = 0;
    int bin = 0;
    for (int b = 255; b >= 0; --b) {
      if (above + h[b] >= static_cast<unsigned>(k)) {
        bin = b;
        break;
      }
      above += h[b];
    }
    sel[0] = static_cast<unsigned>(bin);
    sel[1] = above;
    sel[2] = static_cast<unsigned>(k) - above;
  }
}

__global__ void topk_hist2_kernel(const float* __restrict__ v, int64_t n,
                                  const unsigned* __restrict__ sel1,
                                  unsigned* __restrict__ hist2 /*[256]*/) {
  __shared__ unsigned lh[256];
  if (threadIdx.x < 256) lh[threadIdx.x] = 0;
  __syncthreads();
  const unsigned bin = sel1[0];
  int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (; i < n; i += grid_stride()) {
    const unsigned key = absbits(v[i]);
    if ((key >> 24) == bin) atomicAdd(&lh[(key >> 16) & 0xFFu], 1u);
  }
  __syncthreads();
  if (threadIdx.x < 256 && lh[threadIdx.x])
    atomicAdd(hist2 + threadIdx.x, lh[threadIdx.x]);
}

// select round 2: refine within the chosen top-8-bit bin.
// sel2: [0]=24-bit prefix (bin1<<24 | bin2<<16), [1]=entries still needed
__global__ void topk_select2_kernel(const unsigned* __restrict__ sel1,
                                    const unsigned* __restrict__ hist2,
                                    unsigned* __restrict__ sel2) {
  __shared__ unsigned h[256];
  h[threadIdx.x] = hist2[threadIdx.x];
  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned rem = sel1[2];
    unsigned above = 0;
    int bin = 0;
    for (int b = 255; b >= 0; --b) {
      if (above + h[b] >= rem) {
        bin = b;
        break;
      }
      above += h[b];
    }
    sel2[0] = (sel1[0] << 24) | (static_cast<unsigned>(bin) << 16);
    sel2[1] = rem - above;
  }
}

// round 3: histogram of bits 15..8 within the 16-bit prefix (shrinks the
// candidate bucket ~256x — with k ~ 1e5 the 16-bit bucket alone held
// thousands of entries and the O(c^2) tie ranking exploded).
__global__ void topk_hist3_kernel(const float* __restrict__ v, int64_t n,
                                  const unsigned* __restrict__ sel2,
                                  unsigned* __restrict__ hist3) {
  __shared__ unsigned lh[256];
  if (threadIdx.x < 256) lh[threadIdx.x] = 0;
  __syncthreads();
  const unsigned prefix = sel2[0];
  int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (; i < n; i += grid_stride()) {
    const unsigned key = absbits(v[i]);
    if ((key & 0xFFFF0000u) == prefix)
      atomicAdd(&lh[(key >> 8) & 0xFFu], 1u);
  }
  __syncthreads();
  if (threadIdx.x < 256 && lh[threadIdx.x])
    atomicAdd(hist3 + threadIdx.x, lh[threadIdx.x]);
}

// out2: [0]=24-bit threshold, [1]=need_eq at threshold, [2]=ticket,
// [3]=output cursor
__global__ void topk_select3_kernel(const unsigned* __restrict__ sel2,
                                    const unsigned* __restrict__ hist3,
                                    unsigned* __restrict__ out2) {
  __shared__ unsigned h[256];
  h[threadIdx.x] = hist3[threadIdx.x];
  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned rem = sel2[1];
    unsigned above = 0;
    int bin = 0;
    for (int b = 255; b >= 0; --b) {
      if (above + h[b] >= rem) {
        bin = b;
        break;
      }
      above += h[b];
    }
    out2[0] = sel2[0] | (static_cast<unsigned>(bin) << 8);
    out2[1] = rem - above;
    out2[2] = 0;
    out2[3] = 0;
  }
}

// entries strictly above the 16-bit threshold go straight to the output;
// entries AT the threshold become candidates for the exact final pass.
__global__ void topk_compact_kernel(const float* __restrict__ v, int64_t n,
                                    unsigned* __restrict__ ctl /*out2*/,
                                    int* __restrict__ idx,
                                    float* __restrict__ val, int64_t k,
                                    int* __restrict__ cand, int cand_cap) {
  const unsigned thresh = ctl[0];
  int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (; i < n; i += grid_stride()) {
    const unsigned key = absbits(v[i]) & 0xFFFFFF00u;
    if (key < thresh) continue;
    if (key == thresh) {
      const unsigned c = atomicAdd(&ctl[2], 1u);
      if (c < static_cast<unsigned>(cand_cap)) cand[c] = static_cast<int>(i);
      continue;
    }
    const unsigned pos = atomicAdd(&ctl[3], 1u);
    if (pos < static_cast<unsigned>(k)) {
      idx[pos] = static_cast<int>(i);
      val[pos] = v[i];
    }
  }
}

// exact tie resolution: among the candidates (all sharing the threshold's
// 24-bit key), rank by (|v| desc, index asc) and keep the need_eq best.
// One block; O(c^2/threads) comparisons, c is the candidate count (small:
// one 16-bit bucket of the distribution).
__global__ void topk_ties_kernel(const float* __restrict__ v,
                                 unsigned* __restrict__ ctl,
                                 const int* __restrict__ cand, int cand_cap,
                                 int* __restrict__ idx,
                                 float* __restrict__ val, int64_t k) {
  const unsigned need_eq = ctl[1];
  if (need_eq == 0) return;
  int c = static_cast<int>(ctl[2]);
  if (c > cand_cap) c = cand_cap;
  for (int a = threadIdx.x; a < c; a += blockDim.x) {
    const int ia = cand[a];
    const float va = fabsf(v[ia]);
    unsigned rank = 0;
    for (int b2 = 0; b2 < c; ++b2) {
      const int ib = cand[b2];
      const float vb = fabsf(v[ib]);
      rank += (vb > va) || (vb == va && ib < ia);
    }
    if (rank < need_eq) {
      const unsigned pos = atomicAdd(&ctl[3], 1u);
      if (pos < static_cast<unsigned>(k)) {
        idx[pos] = ia;
        val[pos] = v[ia];
      }
    }
  }
}

__global__ void scatter_add_pairs_kernel(float* __restrict__ out,
                                         const int* __restrict__ idx,
                                         const float* __restrict__ val,
                                         int64_t k) {
  int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (; i < k; i += grid_stride()) atomicAdd(out + idx[i], val[i]);
}

// ---- DP Gaussian noise: device-side philox4x32-10 + Box-Muller ------------
// The Gaussian mechanism's noise is generated ON DEVICE, keyed only on
// (seed, element index): independent of grid shape and rank layout, so
// every rank adds the SAME noise vector to the aggregated delta (the
// simulation-equivalence contract) with no host randn + H2D round trip.
DEV_INLINE unsigned mulhilo32(unsigned a, unsigned b, unsigned* hi) {
  const unsigned long long p = static_cast<unsigned long long>(a) * b;
  *hi = static_cast<unsigned>(p >> 32);
  return static_cast<unsigned>(p);
}

__global__ void philox_gaussian_add_kernel(float* __restrict__ out,
                                           int64_t n,
                                           unsigned long long seed,
                                           float std) {
  const unsigned key0 = static_cast<unsigned>(seed & 0xffffffffull);
  const unsigned key1 = static_cast<unsigned>(seed >> 32);
  const int64_t nq = (n + 3) >> 2;  // one philox block = 4 outputs
  int64_t q = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  for (; q < nq; q += grid_stride()) {
    unsigned c0 = static_cast<unsigned>(q & 0xffffffff);
    unsigned c1 = static_cast<unsigned>(static_cast<unsigned long long>(q)
                                        >> 32);
    unsigned c2 = 0, c3 = 0;
    unsigned k0 = key0, k1 = key1;
#pragma unroll
    for (int r = 0; r < 10; ++r) {
      unsigned hi0, hi1;
      const unsigned lo0 = mulhilo32(0xD2511F53u, c0, &hi0);
      const unsigned lo1 = mulhilo32(0xCD9E8D57u, c2, &hi1);
      const unsigned n0 = hi1 ^ c1 ^ k0;
      const unsigned n2 = hi0 ^ c3 ^ k1;
      c0 = n0;
      c1 = lo1;
      c2 = n2;
      c3 = lo0;
      k0 += 0x9E3779B9u;
      k1 += 0xBB67AE85u;
    }
    // Box-Muller; +0.5 keeps u in (0,1) so log() never sees 0
    constexpr float kInv32 = 2.3283064365386963e-10f;  // 2^-32
    constexpr float kTwoPi = 6.28318530717958647692f;
    const float r1 = sqrtf(-2.f * logf((static_cast<float>(c0) + 0.5f)
                                       * kInv32));
    const float t1 = kTwoPi * (static_cast<float>(c1) * kInv32);
    const float r2 = sqrtf(-2.f * logf((static_cast<float>(c2) + 0.5f)
                                       * kInv32));
    const float t2 = kTwoPi * (static_cast<float>(c3) * kInv32);
    float z[4];
    sincosf(t1, &z[0], &z[1]);
    sincosf(t2, &z[2], &z[3]);
    z[0] *= r1;
    z[1] *= r1;
    z[2] *= r2;
    z[3] *= r2;
    const int64_t base = q << 2;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      if (base + j < n) out[base + j] += std * z[j];
  }
}

// ---- launchers ------------------------------------------------------------
void launch_sq_norm_per_row(const void* g, bool bf16_in, float* sq,
                            float* partials, int64_t rows, int64_t cols,
                            int chunks, hipStream_t stream) {
  dim3 grid(static_cast<unsigned>(rows), chunks);
  if (bf16_in)
    sq_norm_partial_kernel<short><<<grid, 256, 0, stream>>>(
        static_cast<const short*>(g), partials, rows, cols, chunks);
  else
    sq_norm_partial_kernel<float><<<grid, 256, 0, stream>>>(
        static_cast<const float*>(g), partials, rows, cols, chunks);
  sq_norm_combine_kernel<<<(rows + 255) / 256, 256, 0, stream>>>(
      partials, sq, rows, chunks);
}

void launch_scale_rows_mean(const void* g, bool bf16_in,
                            const float* scale, float* out, int64_t rows,
                            int64_t cols, int groups, hipStream_t stream) {
  dim3 grid(elemwise_grid((cols + 3) / 4), 1, groups);
  if (bf16_in)
    scale_rows_mean_kernel<short><<<grid, 256, 0, stream>>>(
        static_cast<const short*>(g), scale, out, rows, cols);
  else
    scale_rows_mean_kernel<float><<<grid, 256, 0, stream>>>(
        static_cast<const float*>(g), scale, out, rows, cols);
}

void launch_topk(const float* v, int64_t n, int64_t k, int* idx, float* val,
                 unsigned* workspace /* >= 1036 unsigned, zeroed */,
                 int* cand, int cand_cap, hipStream_t stream) {
  unsigned* hist1 = workspace;        // 256
  unsigned* hist2 = workspace + 256;  // 256
  unsigned* hist3 = workspace + 512;  // 256
  unsigned* sel1 = workspace + 768;   // 4
  unsigned* sel2 = workspace + 772;   // 4
  unsigned* ctl = workspace + 776;    // 4
  const int grid = elemwise_grid(n);
  topk_hist1_kernel<<<grid, 256, 0, stream>>>(v, n, hist1);
  topk_select_kernel<<<1, 256, 0, stream>>>(hist1, k, sel1);
  topk_hist2_kernel<<<grid, 256, 0, stream>>>(v, n, sel1, hist2);
  topk_select2_kernel<<<1, 256, 0, stream>>>(sel1, hist2, sel2);
  topk_hist3_kernel<<<grid, 256, 0, stream>>>(v, n, sel2, hist3);
  topk_select3_kernel<<<1, 256, 0, stream>>>(sel2, hist3, ctl);
  topk_compact_kernel<<<grid, 256, 0, stream>>>(v, n, ctl, idx, val, k,
                                                cand, cand_cap);
  topk_ties_kernel<<<1, 256, 0, stream>>>(v, ctl, cand, cand_cap, idx, val,
                                          k);
}

void launch_scatter_add_pairs(float* out, const int* idx, const float* val,
                              int64_t k, hipStream_t stream) {
  scatter_add_pairs_kernel<<<elemwise_grid(k), 256, 0, stream>>>(out, idx,
                                                                 val, k);
}

void launch_philox_gaussian_add(float* out, int64_t n,
                                unsigned long long seed, float std,
                                hipStream_t stream) {
  philox_gaussian_add_kernel<<<elemwise_grid((n + 3) / 4), 256, 0, stream>>>(
      out, n, seed, std);
}

}  // namespace colearn
