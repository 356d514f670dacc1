#include "hip/hip_runtime.h"
// bf16 MFMA GEMM, NT layout: C[M,N] = A[M,K] @ B[N,K]^T  (SURVEY.md N5).
//
// The one hot GEMM kernel of the engine: linear forward consumes it
// directly; NN/TN flavors and the attention BMMs route through it with a
// cheap transpose pass (v1 — dedicated transposed-staging kernels are a
// later optimization once rocprof shows the transpose passes matter).
//
// Structure: guide §5 "minimum 2-phase" schedule with register staging
// (T14): issue next tile's global loads BEFORE the MFMA phase so HBM
// latency hides under compute, write them to the other LDS buffer after,
// ONE barrier per K-step, double-buffered LDS, XCD-aware block swizzle.
// Batched via blockIdx.z (strides in elements).

#include "api.h"
#include "tile_gemm.h"

namespace colearn {

// each thread stages 32 B of A and 32 B of B per K-step:
// thread t: row = t>>1, half = t&1 covers k [half*16, half*16+16).
struct StageRegs {
  short8v_t a0, a1, b0, b1;
};

DEV_INLINE void stage_load(const short* __restrict__ A,
                           const short* __restrict__ B, int64_t M, int64_t N,
                           int64_t K, int64_t lda, int64_t ldb, int64_t bm,
                           int64_t bn, int64_t k0, StageRegs& r) {
  const int row = threadIdx.x >> 1;
  const int half = threadIdx.x & 1;
  const int64_t k = k0 + half * 16;
  const int64_t am = bm + row, bn_row = bn + row;
  r.a0 = r.a1 = r.b0 = r.b1 = mfma_frag_zero();
  if (am < M) {
    const short* p = A + am * lda + k;
    if (k + 16 <= K) {
      r.a0 = *reinterpret_cast<const short8v_t*>(p);
      r.a1 = *reinterpret_cast<const short8v_t*>(p + 8);
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j)
        if (k + j < K) {
          if (j < 8) r.a0[j] = p[j]; else r.a1[j - 8] = p[j];
        }
    }
  }
  if (bn_row < N) {
    const short* p = B + bn_row * ldb + k;
    if (k + 16 <= K) {
      r.b0 = *reinterpret_cast<const short8v_t*>(p);
      r.b1 = *reinterpret_cast<const short8v_t*>(p + 8);
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j)
        if (k + j < K) {
          if (j < 8) r.b0[j] = p[j]; else r.b1[j - 8] = p[j];
        }
    }
  }
}

DEV_INLINE void stage_write(short* As, short* Bs, const StageRegs& r) {
  const int row = threadIdx.x >> 1;
  const int half = threadIdx.x & 1;
  *reinterpret_cast<short8v_t*>(reinterpret_cast<char*>(As) +
                                lds_slot_byte(row, half * 2)) = r.a0;
  *reinterpret_cast<short8v_t*>(reinterpret_cast<char*>(As) +
                                lds_slot_byte(row, half * 2 + 1)) = r.a1;
  *reinterpret_cast<short8v_t*>(reinterpret_cast<char*>(Bs) +
                                lds_slot_byte(row, half * 2)) = r.b0;
  *reinterpret_cast<short8v_t*>(reinterpret_cast<char*>(Bs) +
                                lds_slot_byte(row, half * 2 + 1)) = r.b1;
}

__global__ __launch_bounds__(kBlockThreads) void gemm_nt_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    bf16* __restrict__ C, int64_t M, int64_t N, int64_t K, int64_t lda,
    int64_t ldb, int64_t ldc, int64_t sA, int64_t sB, int64_t sC,
    int nbm, int nbn) {
  __shared__ short lds[2][2][BM * BK];  // [dbuf][A|B][tile]
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
  StageRegs regs;
  stage_load(A, B, M, N, K, lda, ldb, bm, bn, 0, regs);
  stage_write(lds[0][0], lds[0][1], regs);
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles)  // issue next tile's loads before compute (T14)
      stage_load(A, B, M, N, K, lda, ldb, bm, bn,
                 static_cast<int64_t>(t + 1) * BK, regs);
    const short* As = lds[cur][0];
    const short* Bs = lds[cur][1];
    short8v_t af[4], bf[4];
#pragma unroll
    for (int m = 0; m < 4; ++m) af[m] = frag_from_lds(As, wm + m * 16);
#pragma unroll
    for (int n = 0; n < 4; ++n) bf[n] = frag_from_lds(Bs, wn + n * 16);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[m], bf[n], acc[m][n], 0, 0, 0);
    if (t + 1 < ntiles) stage_write(lds[cur ^ 1][0], lds[cur ^ 1][1], regs);
    __syncthreads();
    cur ^= 1;
  }
  store_tile_bf16(reinterpret_cast<const float*>(acc), C, ldc, M, N, bm, bn,
                  wm, wn);
}

void launch_gemm_nt(const void* A, const void* B, void* C, int64_t batch,
                    int64_t M, int64_t N, int64_t K, hipStream_t stream) {
  const int nbm = static_cast<int>((M + BM - 1) / BM);
  const int nbn = static_cast<int>((N + BN - 1) / BN);
  dim3 grid(nbm * nbn, 1, static_cast<unsigned>(batch));
 hipLaunchKernelGGL(( gemm_nt_kernel), dim3(grid), dim3(kBlockThreads), 0, stream, 
      static_cast<const short*>(A), static_cast<const short*>(B),
      static_cast<bf16*>(C), M, N, K, K, K, N, M * K, N * K, M * N, nbm,
      nbn);
}

}  // namespace colearn
