// bf16 MFMA GEMM family for gfx950: one kernel template
// covering all four operand layouts, batched.
//
//   C[M,N] = A . B  with A in {[M,K] (K-major), [K,M]} and
//                        B in {[N,K] (K-major), [K,N]}
//
// Both operands stage into canonical K-major LDS tiles.  K-major sources
// stage with 16-B vector loads; K-strided sources load contiguous along
// their leading dim and scatter-transpose into LDS (b16 writes).  This is
// what removes the host-side `.transpose().contiguous()` passes that v1
// used for NN/TN/bmm flavors — measured at 63% of BERT's GEMM time.
//
// Structure: guide §5 "minimum 2-phase" schedule, register staging with
// early load issue (T14), ONE barrier per K-step, double-buffered LDS,
// XOR-swizzled tiles (tile_gemm.h), XCD-aware block swizzle.

#include "api.h"
#include "tile_gemm.h"

namespace colearn {

// GEMM LDS addressing: conv_nt's XOR slot swizzle PLUS a (row>>4) term
// and a physical-row bit6->bit0 remap.  The K-strided operands
// scatter-transpose 16-row groups into the tile (same pattern as conv
// wgrad staging); under the plain scheme those rows collapse onto
// identical banks (16*16 = 0 mod 32) — the extra terms spread them
// while fragment reads (16 CONSECUTIVE rows) stay conflict-free.
// Used consistently for all tile writes and reads in this file.
DEV_INLINE int gemm_slot_byte(int row, int slot) {
  const int pr = row ^ ((row >> 6) & 1);
  return pr * (BK * 2) +
         ((slot ^ ((row >> 2) & 3) ^ ((row >> 4) & 3)) * 16);
}

// byte offset of element (row, k) in the swizzled K-major LDS tile
DEV_INLINE int lds_elem_byte(int row, int k) {
  return gemm_slot_byte(row, k >> 3) + (k & 7) * 2;
}

// fragment read (8 bf16) for this lane; k0 = K-offset (multiple of 32)
DEV_INLINE short8v_t gemm_frag_from_lds(const short* tile, int row_base) {
  const int lane = threadIdx.x % kWave;
  const int row = row_base + (lane & 15);
  const int slot = lane >> 4;
  return *reinterpret_cast<const short8v_t*>(
      reinterpret_cast<const char*>(tile) + gemm_slot_byte(row, slot));
}

// A_KM: A is [M,K] (k contiguous); else A is [K,M] (m contiguous).
// B_KM: B is [N,K] (k contiguous); else B is [K,N] (n contiguous).
template <bool A_KM, bool B_KM, typename OUT = bf16>
__global__ __launch_bounds__(kBlockThreads) void gemm_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    OUT* __restrict__ C, const float* __restrict__ bias, int64_t M,
    int64_t N, int64_t K, int64_t sA, int64_t sB, int64_t sC, int nbm,
    int nbn, int64_t sBias = 0) {
  __shared__ short lds[2][2][BM * BK];
  const int bid = xcd_swizzle(blockIdx.x, nbm * nbn);
  const int64_t bm = static_cast<int64_t>(bid / nbn) * BM;
  const int64_t bn = static_cast<int64_t>(bid % nbn) * BN;
  A += static_cast<int64_t>(blockIdx.z) * sA;
  B += static_cast<int64_t>(blockIdx.z) * sB;
  C += static_cast<int64_t>(blockIdx.z) * sC;

  const int wid = threadIdx.x / kWave;
  const int wm = (wid >> 1) * 64, wn = (wid & 1) * 64;

  f32x4_t acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = static_cast<int>((K + BK - 1) / BK);
  // K-major staging decomposition: thread t = (row = t>>1, khalf = t&1)
  const int row = threadIdx.x >> 1;
  const int half = threadIdx.x & 1;
  // K-strided staging decomposition: thread t = (kk = t>>3, col16 = t&7)
  const int kk = threadIdx.x >> 3;
  const int c16 = (threadIdx.x & 7) * 16;

  short8v_t ra0, ra1, rb0, rb1;

  auto load_km = [&](const short* __restrict__ P, int64_t rows,
                     int64_t k, int64_t r, short8v_t& v0, short8v_t& v1) {
    v0 = v1 = mfma_frag_zero();
    if (r < rows) {
      const short* p = P + r * K + k;
      if (k + 16 <= K) {
        v0 = *reinterpret_cast<const short8v_t*>(p);
        v1 = *reinterpret_cast<const short8v_t*>(p + 8);
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j)
          if (k + j < K) {
            if (j < 8)
              v0[j] = p[j];
            else
              v1[j - 8] = p[j];
          }
      }
    }
  };
  auto load_ks = [&](const short* __restrict__ P, int64_t cols,
                     int64_t k, int64_t c, short8v_t& v0, short8v_t& v1) {
    // P is [K, cols]; load 16 contiguous columns at row k
    v0 = v1 = mfma_frag_zero();
    if (k < K && c < cols) {
      const short* p = P + k * cols + c;
      if (c + 16 <= cols) {
        v0 = *reinterpret_cast<const short8v_t*>(p);
        v1 = *reinterpret_cast<const short8v_t*>(p + 8);
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j)
          if (c + j < cols) {
            if (j < 8)
              v0[j] = p[j];
            else
              v1[j - 8] = p[j];
          }
      }
    }
  };

  auto load_tile = [&](int t) {
    const int64_t k0 = static_cast<int64_t>(t) * BK;
    if constexpr (A_KM)
      load_km(A, M, k0 + half * 16, bm + row, ra0, ra1);
    else
      load_ks(A, M, k0 + kk, bm + c16, ra0, ra1);
    if constexpr (B_KM)
      load_km(B, N, k0 + half * 16, bn + row, rb0, rb1);
    else
      load_ks(B, N, k0 + kk, bn + c16, rb0, rb1);
  };
  auto write_tile = [&](int buf) {
    char* As = reinterpret_cast<char*>(lds[buf][0]);
    char* Bs = reinterpret_cast<char*>(lds[buf][1]);
    if constexpr (A_KM) {
      *reinterpret_cast<short8v_t*>(As + gemm_slot_byte(row, half * 2)) =
          ra0;
      *reinterpret_cast<short8v_t*>(
          As + gemm_slot_byte(row, half * 2 + 1)) = ra1;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        *reinterpret_cast<short*>(As + lds_elem_byte(c16 + j, kk)) = ra0[j];
        *reinterpret_cast<short*>(As + lds_elem_byte(c16 + 8 + j, kk)) =
            ra1[j];
      }
    }
    if constexpr (B_KM) {
      *reinterpret_cast<short8v_t*>(Bs + gemm_slot_byte(row, half * 2)) =
          rb0;
      *reinterpret_cast<short8v_t*>(
          Bs + gemm_slot_byte(row, half * 2 + 1)) = rb1;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        *reinterpret_cast<short*>(Bs + lds_elem_byte(c16 + j, kk)) = rb0[j];
        *reinterpret_cast<short*>(Bs + lds_elem_byte(c16 + 8 + j, kk)) =
            rb1[j];
      }
    }
  };

  load_tile(0);
  write_tile(0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) load_tile(t + 1);
    short8v_t af[4], bfr[4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
      af[m] = gemm_frag_from_lds(lds[cur][0], wm + m * 16);
#pragma unroll
    for (int n = 0; n < 4; ++n)
      bfr[n] = gemm_frag_from_lds(lds[cur][1], wn + n * 16);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[m], bfr[n], acc[m][n], 0, 0, 0);
    if (t + 1 < ntiles) write_tile(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }
  if (bias != nullptr) {
    // fused bias epilogue: one add in registers instead of a whole-output
    // elementwise pass afterwards (sBias: per-batch bias rows, stacked)
    bias += static_cast<int64_t>(blockIdx.z) * sBias;
    const int lane = threadIdx.x % kWave;
    const int cif = lane & 15;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int64_t col = bn + wn + n * 16 + cif;
      const float bv = col < N ? bias[col] : 0.f;
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[m][n][j] += bv;
    }
  }
  store_tile<OUT>(reinterpret_cast<const float*>(acc), C, N, M, N, bm, bn,
                  wm, wn);
}

void launch_gemm(const void* A, const void* B, void* C, const float* bias,
                 int64_t batch, int64_t M, int64_t N, int64_t K, bool a_km,
                 bool b_km, hipStream_t stream, bool f32_out,
                 int64_t sBias) {
  const int nbm = static_cast<int>((M + BM - 1) / BM);
  const int nbn = static_cast<int>((N + BN - 1) / BN);
  dim3 grid(nbm * nbn, 1, static_cast<unsigned>(batch));
  const short* a = static_cast<const short*>(A);
  const short* b = static_cast<const short*>(B);
  const int64_t sA = M * K, sB = N * K, sC = M * N;
  if (f32_out) {
    float* c = static_cast<float*>(C);
    if (a_km && b_km)
      gemm_kernel<true, true, float><<<grid, kBlockThreads, 0, stream>>>(
          a, b, c, bias, M, N, K, sA, sB, sC, nbm, nbn, sBias);
    else if (a_km && !b_km)
      gemm_kernel<true, false, float><<<grid, kBlockThreads, 0, stream>>>(
          a, b, c, bias, M, N, K, sA, sB, sC, nbm, nbn, sBias);
    else if (!a_km && b_km)
      gemm_kernel<false, true, float><<<grid, kBlockThreads, 0, stream>>>(
          a, b, c, bias, M, N, K, sA, sB, sC, nbm, nbn, sBias);
    else
      gemm_kernel<false, false, float><<<grid, kBlockThreads, 0, stream>>>(
          a, b, c, bias, M, N, K, sA, sB, sC, nbm, nbn, sBias);
    return;
  }
  bf16* c = static_cast<bf16*>(C);
  if (a_km && b_km)
    gemm_kernel<true, true><<<grid, kBlockThreads, 0, stream>>>(
        a, b, c, bias, M, N, K, sA, sB, sC, nbm, nbn, sBias);
  else if (a_km && !b_km)
    gemm_kernel<true, false><<<grid, kBlockThreads, 0, stream>>>(
        a, b, c, bias, M, N, K, sA, sB, sC, nbm, nbn, sBias);
  else if (!a_km && b_km)
    gemm_kernel<false, true><<<grid, kBlockThreads, 0, stream>>>(
        a, b, c, bias, M, N, K, sA, sB, sC, nbm, nbn, sBias);
  else
    gemm_kernel<false, false><<<grid, kBlockThreads, 0, stream>>>(
        a, b, c, bias, M, N, K, sA, sB, sC, nbm, nbn, sBias);
}

// legacy entry (conv path compatibility)
void launch_gemm_nt(const void* A, const void* B, void* C, int64_t batch,
                    int64_t M, int64_t N, int64_t K, hipStream_t stream) {
  launch_gemm(A, B, C, nullptr, batch, M, N, K, true, true, stream, false,
              0);
}

}  // namespace colearn
