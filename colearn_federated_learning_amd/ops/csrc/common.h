// Common helpers for the colearn gfx950 (CDNA4) kernels.
//
// Target: AMD Instinct MI355X only (wave64, 256 CUs / 8 XCDs, 160 KiB LDS/CU,
// HBM3E ~8 TB/s).  No CUDA compatibility, no multi-arch dispatch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define DEV_INLINE __device__ __forceinline__

namespace colearn {

constexpr int kWave = 64;           // CDNA wavefront width
constexpr int kNumCU = 256;         // MI355X CUs
constexpr int kNumXCD = 8;          // chiplets, private L2 each

// Grid sizing for memory-bound grid-stride kernels (guide G11):
// cap at ~8 blocks/CU and stride the rest.
DEV_INLINE int64_t grid_stride() {
  return static_cast<int64_t>(blockDim.x) * gridDim.x;
}

inline int elemwise_grid(int64_t n, int block = 256, int max_blocks = 2048) {
  int64_t blocks = (n + block - 1) / block;
  return static_cast<int>(blocks < max_blocks ? blocks : max_blocks);
}

// ---- bf16 <-> f32 ---------------------------------------------------------
using bf16 = __hip_bfloat16;

DEV_INLINE float b2f(bf16 v) { return __bfloat162float(v); }
DEV_INLINE bf16 f2b(float v) { return __float2bfloat16(v); }

// raw-bits views for vectorized bf16 access (guide G13: hipcc does not
// auto-vectorize scalar bf16 loads; 8-16 B per lane is the sweet spot)
struct alignas(16) short8 {
  short x[8];
};
struct alignas(8) short4v {
  short x[4];
};

DEV_INLINE float bits2f(short s) {
  union {
    unsigned u;
    float f;
  } c;
  c.u = static_cast<unsigned>(static_cast<unsigned short>(s)) << 16;
  return c.f;
}
DEV_INLINE short f2bits(float f) {
  // round-to-nearest-even bf16 truncation
  union {
    float f;
    unsigned u;
  } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  unsigned rounded = c.u + 0x7FFFu + lsb;
  return static_cast<short>(rounded >> 16);
}

// ---- wave/block reductions ------------------------------------------------
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  return v;  // valid in lane 0
}

DEV_INLINE float wave_allreduce_sum(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;  // valid in every lane
}

DEV_INLINE float wave_allreduce_max(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// Block reduction via LDS; BLOCK must be a multiple of 64, <= 1024.
// Returns the sum in EVERY thread.  `scratch` needs BLOCK/64 floats.
template <int BLOCK>
DEV_INLINE float block_allreduce_sum(float v, float* scratch) {
  constexpr int kWaves = BLOCK / kWave;
  const int wid = threadIdx.x / kWave;
  v = wave_allreduce_sum(v);
  if (threadIdx.x % kWave == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < kWaves; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

#define HIP_CHECK_LAST()                                          \
  do {                                                            \
    hipError_t e_ = hipGetLastError();                            \
    if (e_ != hipSuccess) {                                       \
      TORCH_CHECK(false, "HIP kernel launch failed: ",            \
                  hipGetErrorString(e_));                         \
    }                                                             \
  } while (0)

}  // namespace colearn
