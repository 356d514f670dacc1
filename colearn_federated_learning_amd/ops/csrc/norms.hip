// Fused LayerNorm (BERT path, N7) and BatchNorm (ResNet path, N6) kernels.
//
// Memory-bound ops: the design targets the HBM roofline with vectorized
// bf16x8 access (guide G13 — scalar bf16 is 2x slower) and single-sweep
// fusion per pass.  BatchNorm operates on CHANNELS-LAST (NHWC) data — the
// layout the whole ResNet GPU path uses — so per-channel statistics
// accumulate with perfectly coalesced lanes (consecutive lanes read
// consecutive channels).  Stats and parameters are fp32.
//
// LayerNorm: one wave per row (row = token), wave shuffle reductions, no
// LDS round-trip for the row stats.

#include "api.h"

#include <cstdlib>
#include "common.h"

namespace colearn {

// Reduction grids are CAPPED at 256 blocks: each block then carries real
// register-accumulation work (large grids degenerate to one iteration per
// thread, where the per-block LDS-atomic flush dominates), and the
// fixed-order combine stays cheap.
int norms_reduce_grid(int64_t n8) { return elemwise_grid(n8, 256, 256); }

// ==== LayerNorm ============================================================
// x: [rows, H] bf16, w/b: [H] fp32, y: bf16, mean/invstd: [rows] fp32
__global__ void layernorm_fwd_kernel(const bf16* __restrict__ x,
                                     const float* __restrict__ wg,
                                     const float* __restrict__ bg,
                                     bf16* __restrict__ y,
                                     float* __restrict__ mean,
                                     float* __restrict__ invstd, int64_t rows,
                                     int H, float eps, int64_t rpg) {
  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int waves = blockDim.x / kWave;
  const float invH = 1.f / H;
  for (int64_t row = static_cast<int64_t>(blockIdx.x) * waves + wid;
       row < rows; row += static_cast<int64_t>(gridDim.x) * waves) {
    const short* xr = reinterpret_cast<const short*>(x + row * H);
    // grouped (stacked-client) weights: rows are client-major
    const float* w = wg + (row / rpg) * H;
    const float* b = bg + (row / rpg) * H;
    float s = 0.f, ss = 0.f;
    for (int h = lane * 8; h + 7 < H; h += kWave * 8) {
      const short8 v = *reinterpret_cast<const short8*>(xr + h);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bits2f(v.x[j]);
        s += f;
        ss += f * f;
      }
    }
    for (int h = (H & ~7) + lane; h < H; h += kWave) {  // H % 8 tail
      const float f = bits2f(xr[h]);
      s += f;
      ss += f * f;
    }
    s = wave_allreduce_sum(s) * invH;
    ss = wave_allreduce_sum(ss) * invH;
    const float istd = rsqrtf(fmaxf(ss - s * s, 0.f) + eps);
    if (lane == 0) {
      mean[row] = s;
      invstd[row] = istd;
    }
    short* yr = reinterpret_cast<short*>(y + row * H);
    for (int h = lane * 8; h + 7 < H; h += kWave * 8) {
      const short8 v = *reinterpret_cast<const short8*>(xr + h);
      const float4 w0 = *reinterpret_cast<const float4*>(w + h);
      const float4 w1 = *reinterpret_cast<const float4*>(w + h + 4);
      const float4 b0 = *reinterpret_cast<const float4*>(b + h);
      const float4 b1 = *reinterpret_cast<const float4*>(b + h + 4);
      const float wv[8] = {w0.x, w0.y, w0.z, w0.w, w1.x, w1.y, w1.z, w1.w};
      const float bv[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.x[j] = f2bits((bits2f(v.x[j]) - s) * istd * wv[j] + bv[j]);
      *reinterpret_cast<short8*>(yr + h) = o;
    }
    for (int h = (H & ~7) + lane; h < H; h += kWave)
      yr[h] = f2bits((bits2f(xr[h]) - s) * istd * w[h] + b[h]);
  }
}

// per-row part of the backward: gx.  (gw/gb accumulate in a column kernel.)
__global__ void layernorm_bwd_gx_kernel(
    const bf16* __restrict__ gy, const bf16* __restrict__ x,
    const float* __restrict__ wg, const float* __restrict__ mean,
    const float* __restrict__ invstd, bf16* __restrict__ gx, int64_t rows,
    int H, int64_t rpg) {
  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int waves = blockDim.x / kWave;
  const float invH = 1.f / H;
  for (int64_t row = static_cast<int64_t>(blockIdx.x) * waves + wid;
       row < rows; row += static_cast<int64_t>(gridDim.x) * waves) {
    const short* gr = reinterpret_cast<const short*>(gy + row * H);
    const short* xr = reinterpret_cast<const short*>(x + row * H);
    const float* w = wg + (row / rpg) * H;
    const float mu = mean[row], istd = invstd[row];
    float s1 = 0.f, s2 = 0.f;  // sum(wg), sum(wg * xhat)
    for (int h = lane; h < H; h += kWave) {
      const float wg = bits2f(gr[h]) * w[h];
      const float xhat = (bits2f(xr[h]) - mu) * istd;
      s1 += wg;
      s2 += wg * xhat;
    }
    s1 = wave_allreduce_sum(s1) * invH;
    s2 = wave_allreduce_sum(s2) * invH;
    short* or_ = reinterpret_cast<short*>(gx + row * H);
    for (int h = lane; h < H; h += kWave) {
      const float wg = bits2f(gr[h]) * w[h];
      const float xhat = (bits2f(xr[h]) - mu) * istd;
      or_[h] = f2bits(istd * (wg - s1 - xhat * s2));
    }
  }
}

// column reduction: gw[h] = sum_rows gy*xhat, gb[h] = sum_rows gy.
// threads along h (coalesced), grid.y chunks the rows; fp32 atomics combine.
__global__ void layernorm_bwd_gwb_kernel(
    const bf16* __restrict__ gy, const bf16* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    float* __restrict__ partials /*[G][gridDim.y][2H]*/, int64_t rpg,
    int H) {
  const int h = blockIdx.x * blockDim.x + threadIdx.x;
  if (h >= H) return;
  // blockIdx.z = group (stacked client); rows client-major
  const int64_t row0 = static_cast<int64_t>(blockIdx.z) * rpg;
  const int64_t rows = row0 + rpg;
  partials += static_cast<int64_t>(blockIdx.z) * gridDim.y * 2 * H;
  float aw = 0.f, ab = 0.f;
  for (int64_t row = row0 + blockIdx.y; row < rows; row += gridDim.y) {
    const float g = bits2f(reinterpret_cast<const short*>(gy)[row * H + h]);
    const float xv = bits2f(reinterpret_cast<const short*>(x)[row * H + h]);
    aw += g * (xv - mean[row]) * invstd[row];
    ab += g;
  }
  float* prow = partials + static_cast<int64_t>(blockIdx.y) * 2 * H;
  prow[h] = aw;
  prow[H + h] = ab;
}

// ==== BatchNorm (channels-last) with fused residual-add + ReLU ===========
// x viewed as [R, C], R = N*H*W; stats per channel.  With 256-thread
// blocks x 8 elems/thread, a thread's channel block is FIXED whenever
// (256*8) % C == 0 (every ResNet channel count) -- partial sums then
// accumulate in REGISTERS and LDS atomics fire once per block instead of
// once per element (the v1 LDS-atomic path measured 31 us on 8 MB; this
// is the fix).  The residual add + ReLU of the ResNet block fuse into the
// normalize/backward sweeps, eliminating the separate elementwise
// kernels and one full activation-tensor round trip per block.

// In-block combine is fully deterministic on the FIXED_C path: each
// thread's 8-channel partials land in a private LDS slot and one thread
// per channel sums its group's slots in fixed order (no atomics anywhere
// in the chain).  The !FIXED_C fallback (C not dividing 2048 — never hit
// by the shipped models) keeps LDS atomics and is only
// tolerance-deterministic.
template <bool FIXED_C>
__global__ void bn_reduce_kernel(const bf16* __restrict__ x, int64_t R,
                                 int C, float* __restrict__ sums) {
  extern __shared__ float lbin[];  // FIXED: [256][16]; else [2C]
  const int64_t n = R * C;
  // stacked clients: grid.z picks the client's activation + partial slab
  x += static_cast<int64_t>(blockIdx.z) * n;
  sums += static_cast<int64_t>(blockIdx.z) * gridDim.x * 2 * C;
  int64_t i = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = grid_stride() * 8;
  if constexpr (FIXED_C) {
    float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    float ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (; i + 7 < n; i += stride) {
      const short8 v = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(x) + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bits2f(v.x[j]);
        s[j] += f;
        ss[j] += f * f;
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      lbin[threadIdx.x * 16 + j] = s[j];
      lbin[threadIdx.x * 16 + 8 + j] = ss[j];
    }
    __syncthreads();
    // channel c is covered by threads tid % (C/8) == c/8, element c%8
    const int nb = C / 8;           // channel blocks
    const int gsz = blockDim.x / nb;  // threads per channel block
    for (int c = threadIdx.x; c < C; c += blockDim.x) {
      const int cb = c / 8, j = c % 8;
      float a = 0.f, b = 0.f;
      for (int k = 0; k < gsz; ++k) {
        a += lbin[(cb + k * nb) * 16 + j];
        b += lbin[(cb + k * nb) * 16 + 8 + j];
      }
      float* out = sums + static_cast<int64_t>(blockIdx.x) * 2 * C;
      out[c] = a;
      out[C + c] = b;
    }
  } else {
    for (int i2 = threadIdx.x; i2 < 2 * C; i2 += blockDim.x) lbin[i2] = 0.f;
    __syncthreads();
    for (; i + 7 < n; i += stride) {
      const short8 v = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(x) + i);
      const int c0 = static_cast<int>(i % C);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bits2f(v.x[j]);
        atomicAdd(&lbin[c0 + j], f);
        atomicAdd(&lbin[C + c0 + j], f * f);
      }
    }
    __syncthreads();
    for (int i2 = threadIdx.x; i2 < 2 * C; i2 += blockDim.x)
      sums[static_cast<int64_t>(blockIdx.x) * 2 * C + i2] = lbin[i2];
  }
}

// combine per-block partial rows (deterministic: fixed lane->entry map +
// fixed shuffle tree).  One wave per output element.
__global__ void partial_combine_kernel(const float* __restrict__ partials,
                                       float* __restrict__ out, int width,
                                       int nblk) {
  partials += static_cast<int64_t>(blockIdx.y) * nblk * width;
  out += static_cast<int64_t>(blockIdx.y) * width;
  const int lane = threadIdx.x % kWave;
  const int i = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  if (i >= width) return;
  float acc = 0.f;
  for (int b = lane; b < nblk; b += kWave)
    acc += partials[static_cast<int64_t>(b) * width + i];
  acc = wave_allreduce_sum(acc);
  if (lane == 0) out[i] = acc;
}

// BN-bwd variant: writes the combined sums into PLANE layout
// [2][G][C] (plane 0 = sum(gym) = gb, plane 1 = sum(gym*xhat) = gw) so
// the binding can return gb/gw as contiguous views — this removes two
// clone kernels per BN backward (40 extra ~4.5us launches per ResNet-18
// step; small-kernel latency was ~20% of step time).
__global__ void partial_combine_planes_kernel(
    const float* __restrict__ partials, float* __restrict__ out_b,
    float* __restrict__ out_w, int C, int nblk) {
  const int width = 2 * C;
  partials += static_cast<int64_t>(blockIdx.y) * nblk * width;
  const int lane = threadIdx.x % kWave;
  const int i = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  if (i >= width) return;
  float acc = 0.f;
  for (int b = lane; b < nblk; b += kWave)
    acc += partials[static_cast<int64_t>(b) * width + i];
  acc = wave_allreduce_sum(acc);
  if (lane == 0) {
    float* out = i < C ? out_b : out_w;
    const int c = i < C ? i : i - C;
    out[static_cast<int64_t>(blockIdx.y) * C + c] = acc;
  }
}

// Pass 2: finalize stats + update running stats over all G*C channel
// instances.  The combined sums are laid out [G][2C]: sum at g*2C + c,
// sumsq at g*2C + C + c.
__global__ void bn_stats_kernel(const float* __restrict__ sums, int C,
                                int G, int64_t R, float* __restrict__ mean,
                                float* __restrict__ invstd,
                                float* __restrict__ running_mean,
                                float* __restrict__ running_var,
                                float momentum, float eps, bool training) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= G * C) return;
  if (training) {
    const int g = i / C, c = i % C;
    const float m = sums[static_cast<int64_t>(g) * 2 * C + c] / R;
    const float var = fmaxf(
        sums[static_cast<int64_t>(g) * 2 * C + C + c] / R - m * m, 0.f);
    mean[i] = m;
    invstd[i] = rsqrtf(var + eps);
    const float unbiased = var * R / fmaxf(static_cast<float>(R - 1), 1.f);
    running_mean[i] = (1.f - momentum) * running_mean[i] + momentum * m;
    running_var[i] = (1.f - momentum) * running_var[i] + momentum * unbiased;
  } else {
    mean[i] = running_mean[i];
    invstd[i] = rsqrtf(running_var[i] + eps);
  }
}

// Pass 3: y = act((x - mean)*invstd*w + b [+ res]), elementwise [R, C].
template <bool RES, bool RELU>
__global__ void bn_normalize_kernel(const bf16* __restrict__ x,
                                    const bf16* __restrict__ res,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ w,
                                    const float* __restrict__ b,
                                    bf16* __restrict__ y, int64_t R, int C) {
  const int64_t n = R * C;
  const int64_t goff = static_cast<int64_t>(blockIdx.z) * n;
  const int gc = blockIdx.z * C;
  x += goff;
  if (RES) res += goff;
  y += goff;
  mean += gc;
  invstd += gc;
  w += gc;
  b += gc;
  int64_t i = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = grid_stride() * 8;
  // stride % C == 0 for every ResNet channel count => per-thread channel
  // block is loop-invariant: hoist the expensive 64-bit modulo + the
  // 4 channel-parameter loads out of the sweep.
  const bool fixed = stride % C == 0;
  if (fixed) {
    const int c0 = static_cast<int>(i % C);
    float mv[8], iv[8], wv[8], bv[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      mv[j] = mean[c0 + j];
      iv[j] = invstd[c0 + j];
      wv[j] = w[c0 + j];
      bv[j] = b[c0 + j];
    }
    for (; i + 7 < n; i += stride) {
      const short8 v = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(x) + i);
      short8 rv;
      if constexpr (RES)
        rv = *reinterpret_cast<const short8*>(
            reinterpret_cast<const short*>(res) + i);
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float r = (bits2f(v.x[j]) - mv[j]) * iv[j] * wv[j] + bv[j];
        if constexpr (RES) r += bits2f(rv.x[j]);
        if constexpr (RELU) r = fmaxf(r, 0.f);
        o.x[j] = f2bits(r);
      }
      *reinterpret_cast<short8*>(reinterpret_cast<short*>(y) + i) = o;
    }
  } else {
    for (; i + 7 < n; i += stride) {
      const short8 v = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(x) + i);
      short8 rv;
      if constexpr (RES)
        rv = *reinterpret_cast<const short8*>(
            reinterpret_cast<const short*>(res) + i);
      const int c0 = static_cast<int>(i % C);
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int c = c0 + j;
        float r = (bits2f(v.x[j]) - mean[c]) * invstd[c] * w[c] + b[c];
        if constexpr (RES) r += bits2f(rv.x[j]);
        if constexpr (RELU) r = fmaxf(r, 0.f);
        o.x[j] = f2bits(r);
      }
      *reinterpret_cast<short8*>(reinterpret_cast<short*>(y) + i) = o;
    }
  }
}

// Backward pass 1: gym = gy * relu_mask(y); gb[c] = sum gym,
// gw[c] = sum gym*xhat; optionally writes gym out (it IS the residual
// grad, and pass 2 consumes it).
template <bool FIXED_C, bool RELU>
__global__ void bn_bwd_reduce_kernel(
    const bf16* __restrict__ gy, const bf16* __restrict__ y,
    const bf16* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, int64_t R, int C,
    float* __restrict__ acc /*[2C]*/, bf16* __restrict__ gym_out) {
  extern __shared__ float lbin[];  // FIXED: [256][16]; else [2C]
  {
    const int64_t goff = static_cast<int64_t>(blockIdx.z) * R * C;
    gy += goff;
    x += goff;
    if (y != nullptr) y += goff;
    if (gym_out != nullptr) gym_out += goff;
    mean += blockIdx.z * C;
    invstd += blockIdx.z * C;
    acc += static_cast<int64_t>(blockIdx.z) * gridDim.x * 2 * C;
  }
  if constexpr (!FIXED_C) {
    for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) lbin[i] = 0.f;
    __syncthreads();
  }
  const int64_t n = R * C;
  int64_t i = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = grid_stride() * 8;
  float ab[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float aw[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const int c0_fixed = static_cast<int>(i % C);
  for (; i + 7 < n; i += stride) {
    const short8 g = *reinterpret_cast<const short8*>(
        reinterpret_cast<const short*>(gy) + i);
    const short8 v = *reinterpret_cast<const short8*>(
        reinterpret_cast<const short*>(x) + i);
    short8 yv;
    if constexpr (RELU)
      yv = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(y) + i);
    const int c0 = FIXED_C ? c0_fixed : static_cast<int>(i % C);
    short8 gm;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c0 + j;
      float gf = bits2f(g.x[j]);
      if constexpr (RELU) gf = bits2f(yv.x[j]) > 0.f ? gf : 0.f;
      gm.x[j] = f2bits(gf);
      const float xh = (bits2f(v.x[j]) - mean[c]) * invstd[c];
      if constexpr (FIXED_C) {
        ab[j] += gf;
        aw[j] += gf * xh;
      } else {
        atomicAdd(&lbin[c], gf);
        atomicAdd(&lbin[C + c], gf * xh);
      }
    }
    if constexpr (RELU)
      *reinterpret_cast<short8*>(reinterpret_cast<short*>(gym_out) + i) = gm;
  }
  if constexpr (FIXED_C) {
    // fixed-order in-block combine (no atomics => deterministic)
    (void)c0_fixed;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      lbin[threadIdx.x * 16 + j] = ab[j];
      lbin[threadIdx.x * 16 + 8 + j] = aw[j];
    }
    __syncthreads();
    const int nb = C / 8;
    const int gsz = blockDim.x / nb;
    for (int c = threadIdx.x; c < C; c += blockDim.x) {
      const int cb = c / 8, j = c % 8;
      float a = 0.f, b2 = 0.f;
      for (int k = 0; k < gsz; ++k) {
        a += lbin[(cb + k * nb) * 16 + j];
        b2 += lbin[(cb + k * nb) * 16 + 8 + j];
      }
      float* out = acc + static_cast<int64_t>(blockIdx.x) * 2 * C;
      out[c] = a;
      out[C + c] = b2;
    }
  } else {
    __syncthreads();
    for (int i2 = threadIdx.x; i2 < 2 * C; i2 += blockDim.x)
      acc[static_cast<int64_t>(blockIdx.x) * 2 * C + i2] = lbin[i2];
  }
}

// Backward pass 2: gx = w*invstd*(gym - gb/n - xhat*gw/n)   (training)
//                  gx = w*invstd*gym                        (eval)
__global__ void bn_bwd_gx_kernel(const bf16* __restrict__ gym,
                                 const bf16* __restrict__ x,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ w,
                                 const float* __restrict__ acc_b,
                                 const float* __restrict__ acc_w,
                                 bf16* __restrict__ gx, int64_t R, int C,
                                 bool training) {
  const int64_t n = R * C;
  const int64_t goff = static_cast<int64_t>(blockIdx.z) * n;
  gym += goff;
  x += goff;
  gx += goff;
  mean += blockIdx.z * C;
  invstd += blockIdx.z * C;
  w += blockIdx.z * C;
  acc_b += static_cast<int64_t>(blockIdx.z) * C;
  acc_w += static_cast<int64_t>(blockIdx.z) * C;
  const float invn = 1.f / R;
  int64_t i = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) * 8;
  const int64_t stride = grid_stride() * 8;
  if (stride % C == 0) {
    // loop-invariant channels: precompute the per-channel coefficients
    // gx = a[j]*gym + (bcoef[j] + d[j]*x)   (training folds the acc terms)
    const int c0 = static_cast<int>(i % C);
    float a[8], bco[8], d[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c0 + j;
      const float wi = w[c] * invstd[c];
      a[j] = wi;
      if (training) {
        const float gwn = acc_w[c] * invn * invstd[c];
        bco[j] = -wi * (acc_b[c] * invn - gwn * mean[c]);
        d[j] = -wi * gwn;
      } else {
        bco[j] = 0.f;
        d[j] = 0.f;
      }
    }
    for (; i + 7 < n; i += stride) {
      const short8 g = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(gym) + i);
      const short8 v = *reinterpret_cast<const short8*>(
          reinterpret_cast<const short*>(x) + i);
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.x[j] = f2bits(a[j] * bits2f(g.x[j]) + bco[j] +
                        d[j] * bits2f(v.x[j]));
      *reinterpret_cast<short8*>(reinterpret_cast<short*>(gx) + i) = o;
    }
    return;
  }
  for (; i + 7 < n; i += stride) {
    const short8 g = *reinterpret_cast<const short8*>(
        reinterpret_cast<const short*>(gym) + i);
    const short8 v = *reinterpret_cast<const short8*>(
        reinterpret_cast<const short*>(x) + i);
    const int c0 = static_cast<int>(i % C);
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c0 + j;
      const float gf = bits2f(g.x[j]);
      float r;
      if (training) {
        const float xhat = (bits2f(v.x[j]) - mean[c]) * invstd[c];
        r = w[c] * invstd[c] *
            (gf - acc_b[c] * invn - xhat * acc_w[c] * invn);
      } else {
        r = w[c] * invstd[c] * gf;
      }
      o.x[j] = f2bits(r);
    }
    *reinterpret_cast<short8*>(reinterpret_cast<short*>(gx) + i) = o;
  }
}

// ---- launchers ------------------------------------------------------------
void launch_layernorm_fwd(const void* x, const float* w, const float* b,
                          void* y, float* mean, float* invstd, int64_t rows,
                          int H, float eps, int G, hipStream_t stream) {
  const int waves = 4;  // 256-thread block, one wave per row
  int64_t g = (rows + waves - 1) / waves;
  const int grid = static_cast<int>(g < 2048 ? g : 2048);
  layernorm_fwd_kernel<<<grid, waves * kWave, 0, stream>>>(
      static_cast<const bf16*>(x), w, b, static_cast<bf16*>(y), mean,
      invstd, rows, H, eps, rows / G);
}

// combine LN gw/gb partials [G][nchunk][2H] straight into the two flat
// grad-slot tensors (no clone, no AccumulateGrad)
__global__ void ln_combine_slots_kernel(const float* __restrict__ partials,
                                        float* __restrict__ gw,
                                        float* __restrict__ gb, int H,
                                        int nblk) {
  const int g = blockIdx.y;
  const float* base =
      partials + static_cast<int64_t>(g) * nblk * 2 * H;
  const int h = blockIdx.x * blockDim.x + threadIdx.x;
  if (h >= H) return;
  float aw = 0.f, ab = 0.f;
  for (int b = 0; b < nblk; ++b) {
    aw += base[static_cast<int64_t>(b) * 2 * H + h];
    ab += base[static_cast<int64_t>(b) * 2 * H + H + h];
  }
  gw[static_cast<int64_t>(g) * H + h] = aw;
  gb[static_cast<int64_t>(g) * H + h] = ab;
}

void launch_ln_combine_slots(const float* partials, float* gw, float* gb,
                             int H, int G, int nchunk,
                             hipStream_t stream) {
  dim3 g2((H + 255) / 256, G);
  ln_combine_slots_kernel<<<g2, 256, 0, stream>>>(partials, gw, gb, H,
                                                  nchunk);
}

void launch_layernorm_bwd(const void* gy_, const void* x_, const float* w,
                          const float* mean, const float* invstd, void* gx_,
                          float* gw_gb /*[G*2H] out*/, float* partials,
                          int64_t rows, int H, int G, hipStream_t stream) {
  const bf16* gy = static_cast<const bf16*>(gy_);
  const bf16* x = static_cast<const bf16*>(x_);
  bf16* gx = static_cast<bf16*>(gx_);
  const int waves = 4;
  int64_t g = (rows + waves - 1) / waves;
  const int grid = static_cast<int>(g < 2048 ? g : 2048);
  layernorm_bwd_gx_kernel<<<grid, waves * kWave, 0, stream>>>(
      gy, x, w, mean, invstd, gx, rows, H, rows / G);
  const int nchunk = 64;
  dim3 g2((H + 255) / 256, nchunk, G);
  layernorm_bwd_gwb_kernel<<<g2, 256, 0, stream>>>(gy, x, mean, invstd,
                                                   partials, rows / G, H);
  partial_combine_kernel<<<dim3((2 * H + 3) / 4, G), 256, 0, stream>>>(
      partials, gw_gb, 2 * H, nchunk);
}

void launch_batchnorm_fwd(const void* x_, const void* res_, const float* w,
                          const float* b, float* running_mean,
                          float* running_var, void* y_, float* mean,
                          float* invstd, float* sums, int64_t R, int C,
                          int G, float momentum, float eps, bool training,
                          bool relu, hipStream_t stream) {
  const bf16* x = static_cast<const bf16*>(x_);
  const bf16* res = static_cast<const bf16*>(res_);
  bf16* y = static_cast<bf16*>(y_);
  const bool fixed = (256 * 8) % C == 0;
  const size_t lds =
      (fixed ? 256 * 16 : 2 * C) * sizeof(float);
  const char* env = getenv("COLEARN_BN_GX_GRID");
  const int gx_cap = env ? atoi(env) : 512;
  const int grid = elemwise_grid(R * C / 8, 256, gx_cap);
  const int rgrid = norms_reduce_grid(R * C / 8);
  if (training) {
    // sums layout: [G][rgrid][2C] partials + [G][2C] combined at the end
    float* partials = sums;
    float* combined =
        sums + static_cast<int64_t>(G) * rgrid * 2 * C;
    if (fixed)
      bn_reduce_kernel<true><<<dim3(rgrid, 1, G), 256, lds, stream>>>(
          x, R, C, partials);
    else
      bn_reduce_kernel<false><<<dim3(rgrid, 1, G), 256, lds, stream>>>(
          x, R, C, partials);
    partial_combine_kernel<<<dim3((2 * C + 3) / 4, G), 256, 0, stream>>>(
        partials, combined, 2 * C, rgrid);
    sums = combined;
  }
  // stats are elementwise over G*C channel instances
  bn_stats_kernel<<<(G * C + 255) / 256, 256, 0, stream>>>(
      sums, C, G, R, mean, invstd, running_mean, running_var, momentum,
      eps, training);
  dim3 ngrid(grid, 1, G);
  if (res != nullptr) {
    if (relu)
      bn_normalize_kernel<true, true><<<ngrid, 256, 0, stream>>>(
          x, res, mean, invstd, w, b, y, R, C);
    else
      bn_normalize_kernel<true, false><<<ngrid, 256, 0, stream>>>(
          x, res, mean, invstd, w, b, y, R, C);
  } else {
    if (relu)
      bn_normalize_kernel<false, true><<<ngrid, 256, 0, stream>>>(
          x, nullptr, mean, invstd, w, b, y, R, C);
    else
      bn_normalize_kernel<false, false><<<ngrid, 256, 0, stream>>>(
          x, nullptr, mean, invstd, w, b, y, R, C);
  }
}

void launch_batchnorm_bwd(const void* gy_, const void* y_, const void* x_,
                          const float* w, const float* mean,
                          const float* invstd, void* gx_, void* gym_,
                          float* acc, float* gwgb_b, float* gwgb_w,
                          int64_t R, int C, int G,
                          bool training, bool relu, hipStream_t stream) {
  const bf16* gy = static_cast<const bf16*>(gy_);
  const bf16* y = static_cast<const bf16*>(y_);
  const bf16* x = static_cast<const bf16*>(x_);
  bf16* gx = static_cast<bf16*>(gx_);
  bf16* gym = static_cast<bf16*>(gym_);
  const bool fixed = (256 * 8) % C == 0;
  const size_t lds =
      (fixed ? 256 * 16 : 2 * C) * sizeof(float);
  const char* env = getenv("COLEARN_BN_GX_GRID");
  const int gx_cap = env ? atoi(env) : 512;
  const int grid = elemwise_grid(R * C / 8, 256, gx_cap);
  const int rgrid = norms_reduce_grid(R * C / 8);
  float* partials = acc;
  acc = partials;
  dim3 rg(rgrid, 1, G);
  if (relu) {
    if (fixed)
      bn_bwd_reduce_kernel<true, true><<<rg, 256, lds, stream>>>(
          gy, y, x, mean, invstd, R, C, acc, gym);
    else
      bn_bwd_reduce_kernel<false, true><<<rg, 256, lds, stream>>>(
          gy, y, x, mean, invstd, R, C, acc, gym);
  } else {
    if (fixed)
      bn_bwd_reduce_kernel<true, false><<<rg, 256, lds, stream>>>(
          gy, y, x, mean, invstd, R, C, acc, nullptr);
    else
      bn_bwd_reduce_kernel<false, false><<<rg, 256, lds, stream>>>(
          gy, y, x, mean, invstd, R, C, acc, nullptr);
  }
  partial_combine_planes_kernel<<<dim3((2 * C + 3) / 4, G), 256, 0,
                                  stream>>>(partials, gwgb_b, gwgb_w, C,
                                            rgrid);
  const bf16* gym_in = relu ? gym : gy;
  bn_bwd_gx_kernel<<<dim3(grid, 1, G), 256, 0, stream>>>(
      gym_in, x, mean, invstd, w, gwgb_b, gwgb_w, gx, R, C, training);
}

}  // namespace colearn
