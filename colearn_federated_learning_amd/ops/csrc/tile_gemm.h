// Shared MFMA tile-GEMM core for gfx950 (guide §5 anatomy).
//
// Geometry: 256-thread block = 4 waves in a 2x2 grid; block tile 128x128;
// per-wave output 64x64 = 4x4 fragments of v_mfma_f32_16x16x32_bf16;
// K-step BK=32 (one MFMA K per step).  LDS tiles are K-major
// ([row][k], 64 B rows) with a 16-B-slot XOR swizzle (slot ^= (row>>2)&3)
// that makes the ds_read_b128 fragment reads bank-conflict-free
// (guide §6 G4: row stride 64 B = 16 banks, 4 rows wrap the bank space,
// so the swizzle must key on row>>2).
//
// Fragment maps (guide §3, 16x16x32 bf16):
//   A: lane l holds A[i = l&15][k = (l>>4)*8 + j], j = 0..7
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C/D: col = lane&15, row = (lane>>4)*4 + reg
#pragma once

#include "common.h"

namespace colearn {

typedef __attribute__((ext_vector_type(8))) short short8v_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 32;
constexpr int kBlockThreads = 256;

// LDS tile: [128 rows][BKT k] bf16; 16-B slots per row with an XOR
// swizzle that keeps every 16-lane ds_read_b128 group conflict-free:
//   BKT=32: 64-B rows (16 banks) -> swz = (row>>2)&3 over 4 slots
//   BKT=64: 128-B rows (32 banks) -> swz = (row>>1)&7 over 8 slots
constexpr int kSlotsPerRow = BK * 2 / 16;  // 4 (BKT=32 default)

template <int BKT = BK>
DEV_INLINE int lds_slot_byte_t(int row, int slot) {
  if constexpr (BKT == 64)
    return row * (BKT * 2) + ((slot ^ ((row >> 1) & 7)) * 16);
  else
    return row * (BKT * 2) + ((slot ^ ((row >> 2) & 3)) * 16);
}

DEV_INLINE int lds_slot_byte(int row, int slot) {
  return lds_slot_byte_t<BK>(row, slot);
}

// read one A/B fragment (8 bf16) for this lane from a K-major LDS tile;
// k0 = K-offset of the fragment within the tile (multiples of 32).
template <int BKT = BK>
DEV_INLINE short8v_t frag_from_lds_t(const short* tile, int row_base,
                                     int k0 = 0) {
  const int lane = threadIdx.x % kWave;
  const int row = row_base + (lane & 15);
  const int slot = (k0 >> 3) + (lane >> 4);  // k-group: k = slot*8 + j
  return *reinterpret_cast<const short8v_t*>(
      reinterpret_cast<const char*>(tile) + lds_slot_byte_t<BKT>(row, slot));
}

DEV_INLINE short8v_t frag_from_lds(const short* tile, int row_base) {
  return frag_from_lds_t<BK>(tile, row_base);
}

DEV_INLINE short8v_t mfma_frag_zero() {
  short8v_t z = {0, 0, 0, 0, 0, 0, 0, 0};
  return z;
}

// Epilogue: write acc[4][4] (each f32x4_t) of one wave into C (bf16 or
// fp32 — OUT=float is the direct-to-flat-grad path), row-major, ldc,
// with bounds checks.  Wave origin (wm, wn) in the block tile; block
// origin (bm, bn) in the matrix.
template <typename OUT>
DEV_INLINE void store_tile(float const* acc_flat /* [4][4][4] */,
                           OUT* __restrict__ c, int64_t ldc, int64_t M,
                           int64_t N, int64_t bm, int64_t bn, int wm,
                           int wn) {
  const int lane = threadIdx.x % kWave;
  const int col_in_frag = lane & 15;
  const int row_grp = lane >> 4;  // rows row_grp*4 + j
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int64_t row = bm + wm + m * 16 + row_grp * 4 + j;
        const int64_t col = bn + wn + n * 16 + col_in_frag;
        if (row < M && col < N) {
          const float v = acc_flat[(m * 4 + n) * 4 + j];
          if constexpr (sizeof(OUT) == 2)
            c[row * ldc + col] = f2b(v);
          else
            c[row * ldc + col] = v;
        }
      }
    }
  }
}

DEV_INLINE void store_tile_bf16(float const* acc_flat, bf16* c, int64_t ldc,
                                int64_t M, int64_t N, int64_t bm,
                                int64_t bn, int wm, int wn) {
  store_tile<bf16>(acc_flat, c, ldc, M, N, bm, bn, wm, wn);
}

// XCD-aware bijective block swizzle (guide T1): contiguous chunks per XCD.
DEV_INLINE int xcd_swizzle(int bid, int nblocks) {
  const int q = nblocks / kNumXCD, r = nblocks % kNumXCD;
  const int xcd = bid % kNumXCD, idx = bid / kNumXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

}  // namespace colearn
