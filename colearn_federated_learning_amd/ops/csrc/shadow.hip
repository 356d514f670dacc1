// Fused shadow-weight sync for the stacked engine (gfx950).
//
// One launch per train step replaces (a) the monolithic fp32->bf16 flat
// cast AND (b) ~20 per-conv weight permute+contiguous copies that the
// dgrad path issued every step (measured 1.1 ms/step of elementwise /
// copy launches — tools/gpu_debug/eager_attrib.py run).
//
// Block ranges:
//   [0, cast_blocks)   : flat fp32 -> bf16 cast, 8 elems/thread
//   [cast_blocks, end) : LDS-tiled transposes producing every conv's
//                        dgrad weight layout [G][CI][R][S][CO] from the
//                        K-major master [G][CO][R][S][CI].
//
// Per-conv segment table (built once in StackedFlatParams.attach_shadows):
//   segA int64 [nseg, 4]: src_off, t_off, blk_start, blk_end
//   segB int32 [nseg, 6]: CO, CI, RS, tilesCO, tilesCI, G
// Each transpose block finds its segment by scanning blk ranges (<=64
// segs, once per block), then moves one 32x32 (co, ci) tile of one
// (g, rs) plane through LDS — both global accesses fully coalesced.

#include "api.h"
#include "common.h"

namespace colearn {

__global__ void shadow_sync_kernel(const float* __restrict__ w,
                                   bf16* __restrict__ shadow, int64_t n,
                                   const int64_t* __restrict__ segA,
                                   const int* __restrict__ segB, int nseg,
                                   bf16* __restrict__ shadow_t,
                                   int cast_blocks) {
  if (blockIdx.x < cast_blocks) {
    // ---- flat cast: 8 elements per thread, grid-stride ----
    const int64_t stride =
        static_cast<int64_t>(cast_blocks) * blockDim.x * 8;
    for (int64_t i = (static_cast<int64_t>(blockIdx.x) * blockDim.x +
                      threadIdx.x) * 8;
         i < n; i += stride) {
      if (i + 7 < n) {
        const float4 a = *reinterpret_cast<const float4*>(w + i);
        const float4 b = *reinterpret_cast<const float4*>(w + i + 4);
        short8 o;
        o.x[0] = f2bits((a.x));
        o.x[1] = f2bits((a.y));
        o.x[2] = f2bits((a.z));
        o.x[3] = f2bits((a.w));
        o.x[4] = f2bits((b.x));
        o.x[5] = f2bits((b.y));
        o.x[6] = f2bits((b.z));
        o.x[7] = f2bits((b.w));
        *reinterpret_cast<short8*>(shadow + i) = o;
      } else {
        for (int64_t j = i; j < n; ++j) shadow[j] = f2b(w[j]);
      }
    }
    return;
  }
  // ---- tiled transpose part ----
  const int b = blockIdx.x - cast_blocks;  // 0-based transpose block id
  int s = 0;
  while (s < nseg && !(segA[s * 4 + 2] <= b && b < segA[s * 4 + 3])) ++s;
  if (s >= nseg) return;
  const int64_t src_off = segA[s * 4 + 0];
  const int64_t t_off = segA[s * 4 + 1];
  const int CO = segB[s * 6 + 0];
  const int CI = segB[s * 6 + 1];
  const int RS = segB[s * 6 + 2];
  const int tCO = segB[s * 6 + 3];
  const int tCI = segB[s * 6 + 4];
  // tile decode: local = (g*tCO + co_t)*tCI + ci_t ; all RS planes are
  // looped INSIDE the block (amortizes the segment scan + launch cost
  // ~9x for 3x3 convs)
  int local = b - static_cast<int>(segA[s * 4 + 2]);
  const int ci_t = local % tCI;
  local /= tCI;
  const int co_t = local % tCO;
  const int g = local / tCO;

  __shared__ float tile[32][33];
  const int tx = threadIdx.x & 31;   // fast dim lane
  const int ty = threadIdx.x >> 5;   // 8 rows per pass
  const int64_t g_src = src_off +
      ((static_cast<int64_t>(g) * CO) * RS) * CI;
  const int64_t g_dst = t_off +
      ((static_cast<int64_t>(g) * CI) * RS) * CO;
  for (int rs = 0; rs < RS; ++rs) {
    // load: rows co, cols ci (ci contiguous in src)
    for (int r = ty; r < 32; r += 8) {
      const int co = co_t * 32 + r;
      const int ci = ci_t * 32 + tx;
      tile[r][tx] =
          w[g_src + (static_cast<int64_t>(co) * RS + rs) * CI + ci];
    }
    __syncthreads();
    // store: rows ci, cols co (co contiguous in dst)
    for (int r = ty; r < 32; r += 8) {
      const int ci = ci_t * 32 + r;
      const int co = co_t * 32 + tx;
      shadow_t[g_dst + (static_cast<int64_t>(ci) * RS + rs) * CO + co] =
          f2b(tile[tx][r]);
    }
    __syncthreads();
  }
}

void launch_shadow_sync(const float* w, void* shadow, int64_t n,
                        const int64_t* segA, const int* segB, int nseg,
                        void* shadow_t, int64_t transpose_blocks,
                        hipStream_t stream) {
  const int cast_blocks = elemwise_grid((n + 7) / 8);
  const int64_t total = cast_blocks + transpose_blocks;
  shadow_sync_kernel<<<static_cast<unsigned>(total), 256, 0, stream>>>(
      w, static_cast<bf16*>(shadow), n, segA, segB, nseg,
      static_cast<bf16*>(shadow_t), cast_blocks);
}

}  // namespace colearn
