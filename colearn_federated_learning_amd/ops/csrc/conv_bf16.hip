// Implicit-GEMM 2-D convolution for gfx950, channels-last bf16
// (ResNet 3x3 & 1x1 convs, fwd + dgrad + wgrad), with
// STACKED-CLIENT execution.
//
// All three passes are MFMA tile GEMMs that GATHER their A operand from
// the NHWC image instead of materializing im2col:
//   fwd   : y[npq, co]  = x_gather[npq, rsci] @ w[co, rsci]^T      (NT)
//   dgrad : gx[nhw, ci] = gy_gather[nhw, rsco] @ wT[ci, rsco]^T    (NT)
//   wgrad : gw[co, rsci] = gy[npq, co]^T @ x_gather[npq, rsci]     (TN,
//           deterministic split-K slabs over npq)
//
// Stacking (the FL multiplexing answer on a 256-CU chip): G virtual
// clients train simultaneously with per-client weights w[G][...] and
// batches stacked along M; grid.z = client.  Each block offsets its
// operand pointers by its client index, so tile structure is unchanged
// while the launch gets G x the workgroups — this is what fills the chip
// for the small deep-layer shapes (measured: ResNet-18 layer4 tiles
// alone = 32 workgroups = 12% occupancy at G=1).
//
// Weights arrive pre-permuted K-major from the wrapper ([G][CO][R][S][CI]
// fwd, [G][CI][R][S][CO] dgrad).  Channel gathers vectorize 8-wide when
// C % 8 == 0 (every ResNet conv except the 3-channel stem).

#include "api.h"

#include <cstdlib>
#include "tile_gemm.h"

namespace colearn {

DEV_INLINE unsigned u32div(unsigned n, U32Div d) {
  return d.m == 0 ? (n >> d.sh) : (__umulhi(n, d.m) >> d.sh);
}

// ---- forward A-gather: row m = (n,p,q), k = (r,s,ci) ---------------------
// The (n,p,q) decomposition costs two 64-bit divisions; hot callers hoist
// it out of their K-loops via conv_fwd_gather_at.
DEV_INLINE void conv_fwd_gather_at(const short* __restrict__ x,
                                   const ConvShape cs, int n, int p, int q,
                                   int64_t k, short8v_t& out) {
  out = mfma_frag_zero();
  const int64_t K_dim = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  if (k >= K_dim) return;
  if ((cs.C & 7) == 0) {
    const unsigned ki = static_cast<unsigned>(k);
    const unsigned kc = u32div(ki, cs.dC);       // k / C
    const int ci = static_cast<int>(ki - kc * cs.C);
    const unsigned r_ = u32div(ki, cs.dSC);      // k / (S*C)
    const int s = static_cast<int>(kc - r_ * cs.S);
    const int r = static_cast<int>(r_);
    const int ih = p * cs.stride - cs.pad + r;
    const int iw = q * cs.stride - cs.pad + s;
    if (ih < 0 || ih >= cs.H || iw < 0 || iw >= cs.W) return;
    out = *reinterpret_cast<const short8v_t*>(
        x + ((static_cast<int64_t>(n) * cs.H + ih) * cs.W + iw) * cs.C + ci);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int64_t kk = k + j;
      if (kk >= K_dim) break;
      const int ci = static_cast<int>(kk % cs.C);
      const int s = static_cast<int>((kk / cs.C) % cs.S);
      const int r =
          static_cast<int>(kk / (static_cast<int64_t>(cs.C) * cs.S));
      const int ih = p * cs.stride - cs.pad + r;
      const int iw = q * cs.stride - cs.pad + s;
      if (ih < 0 || ih >= cs.H || iw < 0 || iw >= cs.W) continue;
      out[j] = x[((static_cast<int64_t>(n) * cs.H + ih) * cs.W + iw) * cs.C +
                 ci];
    }
  }
}

// decompose helper + wrapper keeping the old signature (wgrad's npq
// varies per K-step, so it keeps decomposing)
DEV_INLINE void conv_fwd_gather(const short* __restrict__ x,
                                const ConvShape cs, int64_t m, int64_t k,
                                short8v_t& out) {
  const unsigned mi = static_cast<unsigned>(m);
  const unsigned mq = u32div(mi, cs.dQ);   // m / Q
  const int q = static_cast<int>(mi - mq * cs.Q);
  const unsigned n_ = u32div(mi, cs.dPQ);  // m / (P*Q)
  const int p = static_cast<int>(mq - n_ * cs.P);
  conv_fwd_gather_at(x, cs, static_cast<int>(n_), p, q, k, out);
}

// ---- dgrad A-gather: row m = (n,h,w), k = (r,s,co) -----------------------
// STRIDE1: compile-time fast path without the divisibility checks (most
// ResNet convs are stride 1).
template <bool STRIDE1>
DEV_INLINE void conv_dgrad_gather_at(const short* __restrict__ gy,
                                     const ConvShape cs, int n, int h,
                                     int w, int64_t k, short8v_t& out) {
  out = mfma_frag_zero();
  const int64_t K_dim = static_cast<int64_t>(cs.R) * cs.S * cs.K;
  if (k >= K_dim) return;
  if ((cs.K & 7) == 0) {
    const unsigned ki = static_cast<unsigned>(k);
    const unsigned kk_ = u32div(ki, cs.dK);      // k / K
    const int co = static_cast<int>(ki - kk_ * cs.K);
    const unsigned r_ = u32div(ki, cs.dSK);      // k / (S*K)
    const int s = static_cast<int>(kk_ - r_ * cs.S);
    const int r = static_cast<int>(r_);
    const int ph = h + cs.pad - r;
    const int pw = w + cs.pad - s;
    if (ph < 0 || pw < 0) return;
    int po, qo;
    if constexpr (STRIDE1) {
      po = ph;
      qo = pw;
    } else {
      if (ph % cs.stride || pw % cs.stride) return;
      po = ph / cs.stride;
      qo = pw / cs.stride;
    }
    if (po >= cs.P || qo >= cs.Q) return;
    out = *reinterpret_cast<const short8v_t*>(
        gy + ((static_cast<int64_t>(n) * cs.P + po) * cs.Q + qo) * cs.K + co);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int64_t kk = k + j;
      if (kk >= K_dim) break;
      const int co = static_cast<int>(kk % cs.K);
      const int s = static_cast<int>((kk / cs.K) % cs.S);
      const int r =
          static_cast<int>(kk / (static_cast<int64_t>(cs.K) * cs.S));
      const int ph = h + cs.pad - r;
      const int pw = w + cs.pad - s;
      if (ph < 0 || pw < 0 || ph % cs.stride || pw % cs.stride) continue;
      const int po = ph / cs.stride, qo = pw / cs.stride;
      if (po >= cs.P || qo >= cs.Q) continue;
      out[j] = gy[((static_cast<int64_t>(n) * cs.P + po) * cs.Q + qo) * cs.K +
                  co];
    }
  }
}

template <bool STRIDE1>
DEV_INLINE void conv_dgrad_gather(const short* __restrict__ gy,
                                  const ConvShape cs, int64_t m, int64_t k,
                                  short8v_t& out) {
  const int w = static_cast<int>(m % cs.W);
  const int h = static_cast<int>((m / cs.W) % cs.H);
  const int n = static_cast<int>(m / (static_cast<int64_t>(cs.H) * cs.W));
  conv_dgrad_gather_at<STRIDE1>(gy, cs, n, h, w, k, out);
}

// ---- the shared implicit-GEMM NT kernel ----------------------------------
// GATHER: 0 = conv fwd (gather x), 1 = dgrad stride1, 2 = dgrad strided
// TN: output-channel tile width (64 for the CO=64 layers).
// BKT: K-step depth; 64 halves the barrier count for long-K deep layers.
// TM: M-tile height.  256 is used for the small-N shapes (N <= 64: the
// K=64 layers' fwd and the C=64 layers' dgrad): a 128x64 tile spends as
// many gather instructions per block as MFMA issue; doubling M amortizes
// nothing on the A side (A rows are per-thread) but doubles MFMA per
// B-stage and per barrier, and halves the block count the B operand is
// re-read by.
template <int GATHER, int TM, int TN, int BKT>
__global__ __launch_bounds__(kBlockThreads) void conv_nt_kernel(
    const short* __restrict__ Asrc, const short* __restrict__ Bw,
    bf16* __restrict__ C, ConvShape cs, int64_t Mc /* rows per client */,
    int64_t Nd, int64_t Kd, int64_t a_stride /* A elems per client */,
    int nbm, int nbn) {
  constexpr int QV = BKT / 32;  // 16-element load pairs per thread-row
  constexpr int RPT = TM / 128;  // A staging rows per thread
  __shared__ short lds_a[2][TM * BKT];
  __shared__ short lds_b[2][TN * BKT];
  const int bid = xcd_swizzle(blockIdx.x, nbm * nbn);
  const int64_t bm_local = static_cast<int64_t>(bid / nbn) * TM;
  const int64_t bn = static_cast<int64_t>(bid % nbn) * TN;
  const int g = blockIdx.z;  // stacked client index
  const short* A = Asrc + static_cast<int64_t>(g) * a_stride;
  const short* Bg = Bw + static_cast<int64_t>(g) * Nd * Kd;
  bf16* Cg = C + static_cast<int64_t>(g) * Mc * Nd;

  const int wid = threadIdx.x / kWave;
  constexpr int MF = TM == 256 ? 4 : (TN == 128 ? 4 : 2);
  constexpr int NF = 4;
  const int wm = TM == 256 ? wid * 64
                           : (TN == 128 ? (wid >> 1) * 64 : wid * 32);
  const int wn = (TM != 256 && TN == 128) ? (wid & 1) * 64 : 0;

  f32x4_t acc[MF][NF];
#pragma unroll
  for (int m = 0; m < MF; ++m)
#pragma unroll
    for (int n = 0; n < NF; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  const int row = threadIdx.x >> 1;
  const int half = threadIdx.x & 1;
  const int ntiles = static_cast<int>((Kd + BKT - 1) / BKT);

  // this thread's output pixels are K-loop-invariant: decompose ONCE
  int dn[RPT], dp_[RPT], dq[RPT];
#pragma unroll
  for (int rr = 0; rr < RPT; ++rr) {
    const int64_t am = bm_local + row + rr * 128;
    const int inner = GATHER == 0 ? cs.Q : cs.W;
    const int mid = GATHER == 0 ? cs.P : cs.H;
    dq[rr] = static_cast<int>(am % inner);
    dp_[rr] = static_cast<int>((am / inner) % mid);
    dn[rr] = static_cast<int>(am / (static_cast<int64_t>(mid) * inner));
  }

  short8v_t a0[RPT][2 * QV], b0[2 * QV];
  auto load_tile = [&](int t) {
    const int64_t kbase =
        static_cast<int64_t>(t) * BKT + half * (BKT / 2);
#pragma unroll
    for (int q = 0; q < QV; ++q) {
      const int64_t k = kbase + q * 16;
#pragma unroll
      for (int rr = 0; rr < RPT; ++rr) {
        if (bm_local + row + rr * 128 < Mc) {
          if constexpr (GATHER == 0) {
            conv_fwd_gather_at(A, cs, dn[rr], dp_[rr], dq[rr], k,
                               a0[rr][2 * q]);
            conv_fwd_gather_at(A, cs, dn[rr], dp_[rr], dq[rr], k + 8,
                               a0[rr][2 * q + 1]);
          } else if constexpr (GATHER == 1) {
            conv_dgrad_gather_at<true>(A, cs, dn[rr], dp_[rr], dq[rr], k,
                                       a0[rr][2 * q]);
            conv_dgrad_gather_at<true>(A, cs, dn[rr], dp_[rr], dq[rr],
                                       k + 8, a0[rr][2 * q + 1]);
          } else {
            conv_dgrad_gather_at<false>(A, cs, dn[rr], dp_[rr], dq[rr], k,
                                        a0[rr][2 * q]);
            conv_dgrad_gather_at<false>(A, cs, dn[rr], dp_[rr], dq[rr],
                                        k + 8, a0[rr][2 * q + 1]);
          }
        } else {
          a0[rr][2 * q] = a0[rr][2 * q + 1] = mfma_frag_zero();
        }
      }
      b0[2 * q] = b0[2 * q + 1] = mfma_frag_zero();
      const int64_t bn_row = bn + row;
      if (row < TN && bn_row < Nd) {
        const short* p = Bg + bn_row * Kd + k;
        if (k + 16 <= Kd) {
          b0[2 * q] = *reinterpret_cast<const short8v_t*>(p);
          b0[2 * q + 1] = *reinterpret_cast<const short8v_t*>(p + 8);
        } else {
#pragma unroll
          for (int j = 0; j < 16; ++j)
            if (k + j < Kd) {
              if (j < 8)
                b0[2 * q][j] = p[j];
              else
                b0[2 * q + 1][j - 8] = p[j];
            }
        }
      }
    }
  };
  auto write_tile = [&](int buf) {
    char* As = reinterpret_cast<char*>(lds_a[buf]);
    char* Bs = reinterpret_cast<char*>(lds_b[buf]);
    const int s0 = half * (2 * QV);
#pragma unroll
    for (int q = 0; q < 2 * QV; ++q) {
#pragma unroll
      for (int rr = 0; rr < RPT; ++rr)
        *reinterpret_cast<short8v_t*>(
            As + lds_slot_byte_t<BKT>(row + rr * 128, s0 + q)) = a0[rr][q];
      if (row < TN)
        *reinterpret_cast<short8v_t*>(
            Bs + lds_slot_byte_t<BKT>(row, s0 + q)) = b0[q];
    }
  };

  load_tile(0);
  write_tile(0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) load_tile(t + 1);
#pragma unroll
    for (int kq = 0; kq < QV; ++kq) {
      short8v_t af[MF], bfr[NF];
#pragma unroll
      for (int m = 0; m < MF; ++m)
        af[m] = frag_from_lds_t<BKT>(lds_a[cur], wm + m * 16, kq * 32);
#pragma unroll
      for (int n = 0; n < NF; ++n)
        bfr[n] = frag_from_lds_t<BKT>(lds_b[cur], wn + n * 16, kq * 32);
#pragma unroll
      for (int m = 0; m < MF; ++m)
#pragma unroll
        for (int n = 0; n < NF; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[m], bfr[n], acc[m][n], 0, 0, 0);
    }
    if (t + 1 < ntiles) write_tile(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }
  // bounds-checked epilogue (C/D map: col=lane&15, row=(lane>>4)*4+reg)
  const int lane = threadIdx.x % kWave;
  const int col_in_frag = lane & 15;
  const int row_grp = lane >> 4;
#pragma unroll
  for (int m = 0; m < MF; ++m)
#pragma unroll
    for (int n = 0; n < NF; ++n)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int64_t r = bm_local + wm + m * 16 + row_grp * 4 + j;
        const int64_t c = bn + wn + n * 16 + col_in_frag;
        if (r < Mc && c < Nd)
          Cg[r * Nd + c] = f2b(acc[m][n][j]);
      }
}

// pick the (TN, BKT) instantiation; BKT=64 when the K loop is long.
template <int GATHER>
static void conv_nt_dispatch(const short* A, const short* B, bf16* C,
                             ConvShape cs, int64_t Mc, int64_t Nd,
                             int64_t Kd, int64_t a_stride, int G,
                             hipStream_t stream) {
  const char* env = getenv("COLEARN_CONV_BK");
  // A/B-measured on the flagship bench: BKT=64 wins from Kd ~1k up
  // (fewer barriers per MFMA), 32 wins for shallow-K stem shapes.
  // Strided dgrad (GATHER=2) keeps BKT=32: half its gathers early-exit
  // on the stride check, so deeper staging only adds latency per barrier
  // (measured: stride-2 dgrad kernels ran 84->197 us with BKT=64).
  const int bkt =
      env ? atoi(env) : ((GATHER != 2 && Kd >= 1024) ? 64 : 32);
  if (Nd <= 64) {
    // small-N shapes: a 256x64 tile variant exists (see kernel comment)
    // but measured ~2% SLOWER end-to-end than 128x64 (lower occupancy
    // outweighs the amortized B-staging) — off unless COLEARN_CONV_TM256
    const char* envtm = getenv("COLEARN_CONV_TM256");
    const int nbn = (static_cast<int>(Nd) + 63) / 64;
    const int tm =
        (envtm && atoi(envtm) && Mc >= 256 * 256) ? 256 : 128;
    const int nbm = static_cast<int>((Mc + tm - 1) / tm);
    dim3 grid(nbm * nbn, 1, G);
    if (tm == 256) {
      if (bkt == 64)
        conv_nt_kernel<GATHER, 256, 64, 64>
            <<<grid, kBlockThreads, 0, stream>>>(A, B, C, cs, Mc, Nd, Kd,
                                                 a_stride, nbm, nbn);
      else
        conv_nt_kernel<GATHER, 256, 64, 32>
            <<<grid, kBlockThreads, 0, stream>>>(A, B, C, cs, Mc, Nd, Kd,
                                                 a_stride, nbm, nbn);
    } else if (bkt == 64)
      conv_nt_kernel<GATHER, 128, 64, 64>
          <<<grid, kBlockThreads, 0, stream>>>(A, B, C, cs, Mc, Nd, Kd,
                                               a_stride, nbm, nbn);
    else
      conv_nt_kernel<GATHER, 128, 64, 32>
          <<<grid, kBlockThreads, 0, stream>>>(A, B, C, cs, Mc, Nd, Kd,
                                               a_stride, nbm, nbn);
  } else {
    const int nbn = (static_cast<int>(Nd) + BN - 1) / BN;
    const int nbm = static_cast<int>((Mc + BM - 1) / BM);
    dim3 grid(nbm * nbn, 1, G);
    if (bkt == 64)
      conv_nt_kernel<GATHER, 128, 128, 64>
          <<<grid, kBlockThreads, 0, stream>>>(A, B, C, cs, Mc, Nd, Kd,
                                               a_stride, nbm, nbn);
    else
      conv_nt_kernel<GATHER, 128, 128, 32>
          <<<grid, kBlockThreads, 0, stream>>>(A, B, C, cs, Mc, Nd, Kd,
                                               a_stride, nbm, nbn);
  }
}

// ==== wgrad ===============================================================
// gw[g, co, rsci]: 128x128 tile per block (4 waves, 4x4 fragments each,
// 128 FLOP per staged LDS element — 2x the 64x64 v1), BK=32 over npq;
// deterministic split-K slabs per client; splitk==1 writes the FINAL
// dtype from the epilogue (no slab/combine).  grid = (tiles, splitk, G).
// Both operands transposed-scatter into the XOR-swizzled slot LDS
// (round 2: fragment reads are single conflict-free ds_read_b128s and
// the physical-row remap spreads the scatter-write banks — MFMA-busy
// ratio 0.76 -> 0.96, see profiles/wgrad_r2_evidence.md; the earlier
// ds_read_b64_tr_b16 hardware-transpose plan is superseded).
//
// Tile size is a template param: shapes with K <= 64 (ResNet stem/layer1)
// or RSC <= 64 (1x1 downsamples) were burning >half their MFMA issue on
// tile padding at 128x128 (measured: layer1 tile-occ 0.45) — the 64-wide
// variants restore occupancy.  Waves always cover 64 output rows; the 4
// waves split the other way: TM=128 -> 2x2 waves of 64x(TN/2),
// TM=64 -> 1x4 waves of 64x(TN/4).
template <int TM, int TN, typename OUT = float>
__global__ __launch_bounds__(kBlockThreads) void conv_wgrad_kernel(
    const short* __restrict__ gy, const short* __restrict__ x,
    OUT* __restrict__ gwf, ConvShape cs, int64_t NPQ, int64_t RSC,
    int nbm, int nbn, int splitk) {
  constexpr int WAVES_M = TM / 64;
  constexpr int WAVES_N = 4 / WAVES_M;
  constexpr int WN = TN / WAVES_N;  // wave output cols: 64/32/16
  constexpr int NFR = WN / 16;      // B-fragments per wave
  // XOR-swizzled 16-B slot layout: fragment reads are single
  // conflict-free ds_read_b128s instead of 8 scalar u16 loads — the
  // wgrad loop was issue-bound at 0.76 MFMA-busy ratio on exactly those
  // reads (profiles/pmc_mfma_evidence.md).  The swizzle
  // adds a (row>>4) term on top of conv_nt's (row>>2): the staging
  // writes scatter rows 16 apart (the transpose), which land on
  // identical banks under the plain scheme (64-B rows: 16*16 ≡ 0 mod 32
  // banks) — the extra term spreads them while fragment reads (16
  // CONSECUTIVE rows, so row>>4 constant) stay conflict-free.
  __shared__ short Gs[2][TM * BK];
  __shared__ short Xs[2][TN * BK];
  // physical-row remap (row ^ bit6->bit0) + slot XOR swizzle: together
  // they put all 3 bits of a write lane's 16-row group index into the
  // bank bits, so a staging write cycle's 8 concurrent row-groups hit 8
  // distinct bank groups; fragment reads (16 consecutive logical rows)
  // stay conflict-free — the remap only permutes parity within a group.
  auto wg_slot = [](int row, int slot) {
    const int pr = row ^ ((row >> 6) & 1);
    return pr * (BK * 2) +
           ((slot ^ ((row >> 2) & 3) ^ ((row >> 4) & 3)) * 16);
  };
  const int g = blockIdx.z;
  const short* gyg = gy + static_cast<int64_t>(g) * NPQ * cs.K;
  const short* xg =
      x + static_cast<int64_t>(g) * cs.N * cs.H * cs.W * cs.C;
  OUT* slab = gwf + (static_cast<int64_t>(g) * splitk + blockIdx.y) *
                        cs.K * RSC;
  const int tile = blockIdx.x;
  const int co0 = (tile / nbn) * TM;
  const int rc0 = (tile % nbn) * TN;
  const int64_t chunk = (NPQ + splitk - 1) / splitk;
  const int64_t k_begin = static_cast<int64_t>(blockIdx.y) * chunk;
  const int64_t k_end = min(k_begin + chunk, NPQ);

  const int wid = threadIdx.x / kWave;
  const int wm = (wid / WAVES_N) * 64, wn = (wid % WAVES_N) * WN;
  f32x4_t acc[4][NFR];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < NFR; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  // staging decompositions (per operand: its TM*2 leading threads cover
  // [cols][BK] in 16-element blocks): thread t -> npq row kk, col block.
  // G_ALL/X_ALL are compile-time "every thread participates" facts so the
  // 128-wide instantiations keep branch-free staging (the runtime guard
  // variant cost 62 SGPR spills and 3x on the fully-occupied shapes).
  constexpr bool G_ALL = (TM / 16) * BK >= kBlockThreads;
  constexpr bool X_ALL = (TN / 16) * BK >= kBlockThreads;
  const int kkg = threadIdx.x / (TM / 16);
  const int e16g = (threadIdx.x % (TM / 16)) * 16;
  const int kkx = threadIdx.x / (TN / 16);
  const int e16x = (threadIdx.x % (TN / 16)) * 16;

  short8v_t gv0, gv1, xv0, xv1;
  auto load_gy = [&](int64_t npq) {
    if (co0 + e16g < cs.K) {
      const short* p = gyg + npq * cs.K + co0 + e16g;
      if ((cs.K & 7) == 0 && co0 + e16g + 16 <= cs.K) {
        gv0 = *reinterpret_cast<const short8v_t*>(p);
        gv1 = *reinterpret_cast<const short8v_t*>(p + 8);
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j)
          if (co0 + e16g + j < cs.K) {
            if (j < 8)
              gv0[j] = p[j];
            else
              gv1[j - 8] = p[j];
          }
      }
    }
  };
  auto load_chunk = [&](int64_t k0) {
    gv0 = gv1 = xv0 = xv1 = mfma_frag_zero();
    if constexpr (TM == TN && G_ALL) {
      // equal tiles, every thread stages both operands from the same npq
      // row: single guard, merged branch (matches the pre-template
      // codegen — the split-guard form spilled 53 SGPRs and ran 3x slower
      // on the fully-occupied shapes)
      const int64_t npq = k0 + kkg;
      if (npq < k_end) {
        load_gy(npq);
        conv_fwd_gather(xg, cs, npq, rc0 + e16x, xv0);
        conv_fwd_gather(xg, cs, npq, rc0 + e16x + 8, xv1);
      }
    } else {
      if ((G_ALL || kkg < BK) && k0 + kkg < k_end) load_gy(k0 + kkg);
      if ((X_ALL || kkx < BK) && k0 + kkx < k_end) {
        const int64_t npq = k0 + kkx;
        conv_fwd_gather(xg, cs, npq, rc0 + e16x, xv0);
        conv_fwd_gather(xg, cs, npq, rc0 + e16x + 8, xv1);
      }
    }
  };
  auto write_chunk = [&](int buf) {
    if (G_ALL || kkg < BK) {
      const int slot_g = kkg >> 3, sub_g = (kkg & 7) * 2;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        *reinterpret_cast<short*>(
            reinterpret_cast<char*>(Gs[buf]) +
            wg_slot(e16g + j, slot_g) + sub_g) = gv0[j];
        *reinterpret_cast<short*>(
            reinterpret_cast<char*>(Gs[buf]) +
            wg_slot(e16g + 8 + j, slot_g) + sub_g) = gv1[j];
      }
    }
    if (X_ALL || kkx < BK) {
      const int slot_x = kkx >> 3, sub_x = (kkx & 7) * 2;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        *reinterpret_cast<short*>(
            reinterpret_cast<char*>(Xs[buf]) +
            wg_slot(e16x + j, slot_x) + sub_x) = xv0[j];
        *reinterpret_cast<short*>(
            reinterpret_cast<char*>(Xs[buf]) +
            wg_slot(e16x + 8 + j, slot_x) + sub_x) = xv1[j];
      }
    }
  };

  // double-buffered pipeline: next chunk's global loads issue before the
  // MFMA phase (HBM latency hides under compute — T14), LDS write lands
  // in the other buffer, ONE barrier per K-step.
  load_chunk(k_begin);
  write_chunk(0);
  __syncthreads();
  int cur = 0;
  for (int64_t k0 = k_begin; k0 < k_end; k0 += BK) {
    const bool has_next = k0 + BK < k_end;
    if (has_next) load_chunk(k0 + BK);
    const int lane = threadIdx.x % kWave;
    const int fr = lane & 15, slot = lane >> 4;
    short8v_t af[4], bfr[NFR];
#pragma unroll
    for (int m = 0; m < 4; ++m)
      af[m] = *reinterpret_cast<const short8v_t*>(
          reinterpret_cast<const char*>(Gs[cur]) +
          wg_slot(wm + m * 16 + fr, slot));
#pragma unroll
    for (int n = 0; n < NFR; ++n)
      bfr[n] = *reinterpret_cast<const short8v_t*>(
          reinterpret_cast<const char*>(Xs[cur]) +
          wg_slot(wn + n * 16 + fr, slot));
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < NFR; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[m], bfr[n], acc[m][n], 0, 0, 0);
    if (has_next) write_chunk(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }
  // deterministic slab epilogue
  const int lane = threadIdx.x % kWave;
  const int cif = lane & 15, rg = lane >> 4;
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < NFR; ++n)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int co = co0 + wm + m * 16 + rg * 4 + j;
        const int rc = rc0 + wn + n * 16 + cif;
        if (co < cs.K && rc < RSC) {
          if constexpr (sizeof(OUT) == 2)
            slab[static_cast<int64_t>(co) * RSC + rc] = f2b(acc[m][n][j]);
          else
            slab[static_cast<int64_t>(co) * RSC + rc] = acc[m][n][j];
        }
      }
}

// sum the split-K slabs of each client in slice order; OUT is bf16 for
// the autocast grad path or float for the shadow-weight path (grads land
// directly in the fp32 flat grad layout, no bf16 round trip).
template <typename OUT>
__global__ void wgrad_combine_kernel(const float* __restrict__ gwf,
                                     OUT* __restrict__ gw, int64_t n_per_g,
                                     int splitk) {
  const int g = blockIdx.y;
  const float* base = gwf + static_cast<int64_t>(g) * splitk * n_per_g;
  OUT* out = gw + static_cast<int64_t>(g) * n_per_g;
  int64_t i =
      (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = static_cast<int64_t>(blockDim.x) * gridDim.x * 4;
  for (; i + 3 < n_per_g; i += stride) {
    float4 a = *reinterpret_cast<const float4*>(base + i);
    for (int sk = 1; sk < splitk; ++sk) {
      const float4 v =
          *reinterpret_cast<const float4*>(base + sk * n_per_g + i);
      a.x += v.x;
      a.y += v.y;
      a.z += v.z;
      a.w += v.w;
    }
    if constexpr (sizeof(OUT) == 2) {
      short4v o;
      o.x[0] = f2bits(a.x);
      o.x[1] = f2bits(a.y);
      o.x[2] = f2bits(a.z);
      o.x[3] = f2bits(a.w);
      *reinterpret_cast<short4v*>(reinterpret_cast<short*>(out) + i) = o;
    } else {
      *reinterpret_cast<float4*>(reinterpret_cast<float*>(out) + i) = a;
    }
  }
  if (blockIdx.x == 0 && threadIdx.x < (n_per_g & 3)) {
    const int64_t j = (n_per_g & ~int64_t(3)) + threadIdx.x;
    float acc = 0.f;
    for (int sk = 0; sk < splitk; ++sk) acc += base[sk * n_per_g + j];
    if constexpr (sizeof(OUT) == 2)
      out[j] = f2b(acc);
    else
      out[j] = acc;
  }
}

// ---- launchers ------------------------------------------------------------
void launch_conv_fwd(const void* x, const void* w, void* y, ConvShape cs,
                     int G, hipStream_t stream) {
  const int64_t Mc = static_cast<int64_t>(cs.N) * cs.P * cs.Q;
  const int64_t Kd = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  const int64_t a_stride = static_cast<int64_t>(cs.N) * cs.H * cs.W * cs.C;
  conv_nt_dispatch<0>(static_cast<const short*>(x),
                      static_cast<const short*>(w), static_cast<bf16*>(y),
                      cs, Mc, cs.K, Kd, a_stride, G, stream);
}

void launch_conv_dgrad(const void* gy, const void* wt, void* gx,
                       ConvShape cs, int G, hipStream_t stream) {
  const int64_t Mc = static_cast<int64_t>(cs.N) * cs.H * cs.W;
  const int64_t Kd = static_cast<int64_t>(cs.R) * cs.S * cs.K;
  const int64_t a_stride = static_cast<int64_t>(cs.N) * cs.P * cs.Q * cs.K;
  if (cs.stride == 1)
    conv_nt_dispatch<1>(static_cast<const short*>(gy),
                        static_cast<const short*>(wt),
                        static_cast<bf16*>(gx), cs, Mc, cs.C, Kd, a_stride,
                        G, stream);
  else
    conv_nt_dispatch<2>(static_cast<const short*>(gy),
                        static_cast<const short*>(wt),
                        static_cast<bf16*>(gx), cs, Mc, cs.C, Kd, a_stride,
                        G, stream);
}

// tile selection: small K / small RSC shapes use 64-wide tiles so the
// padding waste stays bounded (see kernel comment)
static void wgrad_tiles(ConvShape cs, int64_t RSC, int& tm, int& tn,
                        int& nbm, int& nbn) {
  tm = cs.K <= 64 ? 64 : 128;
  tn = RSC <= 64 ? 64 : 128;
  nbm = (cs.K + tm - 1) / tm;
  nbn = static_cast<int>((RSC + tn - 1) / tn);
}

int conv_wgrad_splitk(ConvShape cs, int G) {
  const int64_t NPQ = static_cast<int64_t>(cs.N) * cs.P * cs.Q;
  const int64_t RSC = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  int tm, tn, nbm, nbn;
  wgrad_tiles(cs, RSC, tm, tn, nbm, nbn);
  int64_t want = 512 / (static_cast<int64_t>(nbm) * nbn * G);
  if (want < 1) want = 1;
  int64_t cap = (NPQ + BK - 1) / BK;
  return static_cast<int>(want < cap ? want : cap);
}

template <typename OUT>
static void wgrad_dispatch(const short* gy, const short* x, OUT* gwf,
                           ConvShape cs, int64_t NPQ, int64_t RSC, int G,
                           int splitk, hipStream_t stream) {
  int tm, tn, nbm, nbn;
  wgrad_tiles(cs, RSC, tm, tn, nbm, nbn);
  dim3 grid(nbm * nbn, splitk, G);
  if (tm == 64 && tn == 64)
    conv_wgrad_kernel<64, 64, OUT><<<grid, kBlockThreads, 0, stream>>>(
        gy, x, gwf, cs, NPQ, RSC, nbm, nbn, splitk);
  else if (tm == 64)
    conv_wgrad_kernel<64, 128, OUT><<<grid, kBlockThreads, 0, stream>>>(
        gy, x, gwf, cs, NPQ, RSC, nbm, nbn, splitk);
  else if (tn == 64)
    conv_wgrad_kernel<128, 64, OUT><<<grid, kBlockThreads, 0, stream>>>(
        gy, x, gwf, cs, NPQ, RSC, nbm, nbn, splitk);
  else
    conv_wgrad_kernel<128, 128, OUT><<<grid, kBlockThreads, 0, stream>>>(
        gy, x, gwf, cs, NPQ, RSC, nbm, nbn, splitk);
}

void launch_conv_wgrad_nosum(const void* gy, const void* x, float* gwf,
                             ConvShape cs, int G, hipStream_t stream) {
  const int64_t NPQ = static_cast<int64_t>(cs.N) * cs.P * cs.Q;
  const int64_t RSC = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  wgrad_dispatch<float>(static_cast<const short*>(gy),
                        static_cast<const short*>(x), gwf, cs, NPQ, RSC,
                        G, 1, stream);
}

void launch_conv_wgrad(const void* gy, const void* x, float* gwf, void* gw,
                       ConvShape cs, int G, hipStream_t stream) {
  launch_conv_wgrad_out(gy, x, gwf, gw, cs, G, false, stream);
}

void launch_conv_wgrad_out(const void* gy, const void* x, float* gwf,
                           void* gw, ConvShape cs, int G, bool f32_out,
                           hipStream_t stream) {
  const int64_t NPQ = static_cast<int64_t>(cs.N) * cs.P * cs.Q;
  const int64_t RSC = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  const int splitk = conv_wgrad_splitk(cs, G);
  if (splitk == 1) {
    // no K-split: the epilogue writes the FINAL output dtype straight
    // into gw — no fp32 slab, no combine pass (the per-sample DP path
    // at G=4096 wrote ~180 GB/step of slabs just to cast them)
    if (f32_out)
      wgrad_dispatch<float>(static_cast<const short*>(gy),
                            static_cast<const short*>(x),
                            static_cast<float*>(gw), cs, NPQ, RSC, G, 1,
                            stream);
    else
      wgrad_dispatch<bf16>(static_cast<const short*>(gy),
                           static_cast<const short*>(x),
                           static_cast<bf16*>(gw), cs, NPQ, RSC, G, 1,
                           stream);
    return;
  }
  wgrad_dispatch<float>(static_cast<const short*>(gy),
                        static_cast<const short*>(x), gwf, cs, NPQ, RSC,
                        G, splitk, stream);
  const int64_t n_per_g = static_cast<int64_t>(cs.K) * RSC;
  // target ~8-16k total blocks: the per-sample-grad path launches with
  // G in the thousands (the old 2048/G cap collapsed to ONE block per g
  // and ran the combine at 0.5 TB/s)
  int cap = G > 0 ? 16384 / G : 2048;
  if (cap < 1) cap = 1;
  if (cap > 2048) cap = 2048;
  int cgrid_x = elemwise_grid((n_per_g + 3) / 4, 256, cap);
  if (cgrid_x < 1) cgrid_x = 1;
  if (f32_out)
    wgrad_combine_kernel<<<dim3(cgrid_x, G), 256, 0, stream>>>(
        gwf, static_cast<float*>(gw), n_per_g, splitk);
  else
    wgrad_combine_kernel<<<dim3(cgrid_x, G), 256, 0, stream>>>(
        gwf, static_cast<bf16*>(gw), n_per_g, splitk);
}

// ==== wgrad layout variants (tuning bench only; prod kernel above) =======
// V1: conv-tile layout (64-B rows, 16-B slot XOR swizzle) — scatter b16
//     writes, proven conflict-free b128 fragment reads.
// V2: natural [k][col] tile — vector b128 writes, strided b16 reads,
//     +8-short row pad to spread read banks.
__global__ __launch_bounds__(kBlockThreads) void conv_wgrad_v1_kernel(
    const short* __restrict__ gy, const short* __restrict__ x,
    float* __restrict__ gwf, ConvShape cs, int64_t NPQ, int64_t RSC,
    int nbm, int nbn, int splitk) {
  constexpr int TM = 128, TN = 128;
  __shared__ short Gs[2][TM * BK];
  __shared__ short Xs[2][TN * BK];
  const int g = blockIdx.z;
  const short* gyg = gy + static_cast<int64_t>(g) * NPQ * cs.K;
  const short* xg = x + static_cast<int64_t>(g) * cs.N * cs.H * cs.W * cs.C;
  float* slab = gwf + (static_cast<int64_t>(g) * splitk + blockIdx.y) *
                          cs.K * RSC;
  const int co0 = (blockIdx.x / nbn) * TM;
  const int rc0 = (blockIdx.x % nbn) * TN;
  const int64_t chunk = (NPQ + splitk - 1) / splitk;
  const int64_t k_begin = static_cast<int64_t>(blockIdx.y) * chunk;
  const int64_t k_end = min(k_begin + chunk, NPQ);
  const int wid = threadIdx.x / kWave;
  const int wm = (wid >> 1) * 64, wn = (wid & 1) * 64;
  f32x4_t acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};
  const int kk = threadIdx.x >> 3;
  const int e16 = (threadIdx.x & 7) * 16;
  short8v_t gv0, gv1, xv0, xv1;
  auto load_chunk = [&](int64_t k0) {
    const int64_t npq = k0 + kk;
    gv0 = gv1 = xv0 = xv1 = mfma_frag_zero();
    if (npq < k_end) {
      if (co0 + e16 < cs.K) {
        const short* p = gyg + npq * cs.K + co0 + e16;
        if ((cs.K & 7) == 0 && co0 + e16 + 16 <= cs.K) {
          gv0 = *reinterpret_cast<const short8v_t*>(p);
          gv1 = *reinterpret_cast<const short8v_t*>(p + 8);
        }
      }
      conv_fwd_gather(xg, cs, npq, rc0 + e16, xv0);
      conv_fwd_gather(xg, cs, npq, rc0 + e16 + 8, xv1);
    }
  };
  auto write_chunk = [&](int buf) {
    char* G = reinterpret_cast<char*>(Gs[buf]);
    char* X = reinterpret_cast<char*>(Xs[buf]);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *reinterpret_cast<short*>(
          G + lds_slot_byte(e16 + j, kk >> 3) + (kk & 7) * 2) = gv0[j];
      *reinterpret_cast<short*>(
          G + lds_slot_byte(e16 + 8 + j, kk >> 3) + (kk & 7) * 2) = gv1[j];
      *reinterpret_cast<short*>(
          X + lds_slot_byte(e16 + j, kk >> 3) + (kk & 7) * 2) = xv0[j];
      *reinterpret_cast<short*>(
          X + lds_slot_byte(e16 + 8 + j, kk >> 3) + (kk & 7) * 2) = xv1[j];
    }
  };
  load_chunk(k_begin);
  write_chunk(0);
  __syncthreads();
  int cur = 0;
  for (int64_t k0 = k_begin; k0 < k_end; k0 += BK) {
    const bool has_next = k0 + BK < k_end;
    if (has_next) load_chunk(k0 + BK);
    short8v_t af[4], bfr[4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
      af[m] = frag_from_lds(Gs[cur], wm + m * 16);
#pragma unroll
    for (int n = 0; n < 4; ++n)
      bfr[n] = frag_from_lds(Xs[cur], wn + n * 16);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[m], bfr[n], acc[m][n], 0, 0, 0);
    if (has_next) write_chunk(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }
  const int lane = threadIdx.x % kWave;
  const int cif = lane & 15, rg = lane >> 4;
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int co = co0 + wm + m * 16 + rg * 4 + j;
        const int rc = rc0 + wn + n * 16 + cif;
        if (co < cs.K && rc < RSC)
          slab[static_cast<int64_t>(co) * RSC + rc] = acc[m][n][j];
      }
}

__global__ __launch_bounds__(kBlockThreads) void conv_wgrad_v2_kernel(
    const short* __restrict__ gy, const short* __restrict__ x,
    float* __restrict__ gwf, ConvShape cs, int64_t NPQ, int64_t RSC,
    int nbm, int nbn, int splitk) {
  constexpr int TM = 128, TN = 128, PAD = 8;
  __shared__ short Gs[2][BK][TM + PAD];
  __shared__ short Xs[2][BK][TN + PAD];
  const int g = blockIdx.z;
  const short* gyg = gy + static_cast<int64_t>(g) * NPQ * cs.K;
  const short* xg = x + static_cast<int64_t>(g) * cs.N * cs.H * cs.W * cs.C;
  float* slab = gwf + (static_cast<int64_t>(g) * splitk + blockIdx.y) *
                          cs.K * RSC;
  const int co0 = (blockIdx.x / nbn) * TM;
  const int rc0 = (blockIdx.x % nbn) * TN;
  const int64_t chunk = (NPQ + splitk - 1) / splitk;
  const int64_t k_begin = static_cast<int64_t>(blockIdx.y) * chunk;
  const int64_t k_end = min(k_begin + chunk, NPQ);
  const int wid = threadIdx.x / kWave;
  const int wm = (wid >> 1) * 64, wn = (wid & 1) * 64;
  f32x4_t acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};
  const int kk = threadIdx.x >> 3;
  const int e16 = (threadIdx.x & 7) * 16;
  short8v_t gv0, gv1, xv0, xv1;
  auto load_chunk = [&](int64_t k0) {
    const int64_t npq = k0 + kk;
    gv0 = gv1 = xv0 = xv1 = mfma_frag_zero();
    if (npq < k_end) {
      if (co0 + e16 < cs.K) {
        const short* p = gyg + npq * cs.K + co0 + e16;
        if ((cs.K & 7) == 0 && co0 + e16 + 16 <= cs.K) {
          gv0 = *reinterpret_cast<const short8v_t*>(p);
          gv1 = *reinterpret_cast<const short8v_t*>(p + 8);
        }
      }
      conv_fwd_gather(xg, cs, npq, rc0 + e16, xv0);
      conv_fwd_gather(xg, cs, npq, rc0 + e16 + 8, xv1);
    }
  };
  auto write_chunk = [&](int buf) {
    *reinterpret_cast<short8v_t*>(&Gs[buf][kk][e16]) = gv0;
    *reinterpret_cast<short8v_t*>(&Gs[buf][kk][e16 + 8]) = gv1;
    *reinterpret_cast<short8v_t*>(&Xs[buf][kk][e16]) = xv0;
    *reinterpret_cast<short8v_t*>(&Xs[buf][kk][e16 + 8]) = xv1;
  };
  load_chunk(k_begin);
  write_chunk(0);
  __syncthreads();
  int cur = 0;
  const int lane = threadIdx.x % kWave;
  const int fr = lane & 15, kg = (lane >> 4) * 8;
  for (int64_t k0 = k_begin; k0 < k_end; k0 += BK) {
    const bool has_next = k0 + BK < k_end;
    if (has_next) load_chunk(k0 + BK);
    short8v_t af[4], bfr[4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int j = 0; j < 8; ++j) af[m][j] = Gs[cur][kg + j][wm + m * 16 + fr];
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int j = 0; j < 8; ++j) bfr[n][j] = Xs[cur][kg + j][wn + n * 16 + fr];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[m], bfr[n], acc[m][n], 0, 0, 0);
    if (has_next) write_chunk(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }
  const int cif = lane & 15, rg = lane >> 4;
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int co = co0 + wm + m * 16 + rg * 4 + j;
        const int rc = rc0 + wn + n * 16 + cif;
        if (co < cs.K && rc < RSC)
          slab[static_cast<int64_t>(co) * RSC + rc] = acc[m][n][j];
      }
}

void launch_conv_wgrad_variant(const void* gy, const void* x, float* gwf,
                               void* gw, ConvShape cs, int G, int variant,
                               hipStream_t stream) {
  const int64_t NPQ = static_cast<int64_t>(cs.N) * cs.P * cs.Q;
  const int64_t RSC = static_cast<int64_t>(cs.R) * cs.S * cs.C;
  const int nbm = (cs.K + 127) / 128;
  const int nbn = static_cast<int>((RSC + 127) / 128);
  const int splitk = conv_wgrad_splitk(cs, G);
  dim3 grid(nbm * nbn, splitk, G);
  if (variant == 1)
    conv_wgrad_v1_kernel<<<grid, kBlockThreads, 0, stream>>>(
        static_cast<const short*>(gy), static_cast<const short*>(x), gwf,
        cs, NPQ, RSC, nbm, nbn, splitk);
  else
    conv_wgrad_v2_kernel<<<grid, kBlockThreads, 0, stream>>>(
        static_cast<const short*>(gy), static_cast<const short*>(x), gwf,
        cs, NPQ, RSC, nbm, nbn, splitk);
  const int64_t n_per_g = static_cast<int64_t>(cs.K) * RSC;
  // target ~8-16k total blocks: the per-sample-grad path launches with
  // G in the thousands (the old 2048/G cap collapsed to ONE block per g
  // and ran the combine at 0.5 TB/s)
  int cap = G > 0 ? 16384 / G : 2048;
  if (cap < 1) cap = 1;
  if (cap > 2048) cap = 2048;
  int cgrid_x = elemwise_grid((n_per_g + 3) / 4, 256, cap);
  if (cgrid_x < 1) cgrid_x = 1;
  wgrad_combine_kernel<<<dim3(cgrid_x, G), 256, 0, stream>>>(
      gwf, static_cast<bf16*>(gw), n_per_g, splitk);
}

// ---- ds_read_b64_tr_b16 mapping probe (tools/gpu_debug) -----------------
// One wave; LDS filled with its own linear index; each lane reads 8 B at
// addr = lane*8 (+scheme variations); dump shows which LDS element lands
// in (lane, j) so the wgrad staging image can be designed for the
// hardware transpose read.
// STATUS: the instruction assembles (dest VGPR pair + per-lane address)
// but returned zeros for all schemes on hardware — the operand/addressing
// contract needs more probing before wgrad staging can use it.  Parked as
// the documented next wgrad step (SURVEY.md deferred items).
typedef __attribute__((ext_vector_type(4))) short short4v_probe;
__global__ void tr_probe_kernel(short* __restrict__ out, int scheme) {
  __shared__ short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += 64)
    lds[i] = static_cast<short>(i);
  __syncthreads();
  const int l = threadIdx.x;
  unsigned addr;
  if (scheme == 0)
    addr = l * 8;  // plain per-lane contiguous
  else if (scheme == 1)
    addr = ((l & 15) * 4 + (l >> 4) * 256) * 2;  // 4-col groups per lane
  else
    addr = ((l >> 4) * 4 + (l & 15) * 32) * 2;   // k-group-major
  short4v_probe v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v)
               : "v"(addr));
#pragma unroll
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = v[j];
}

void launch_tr_probe(void* out, int scheme, hipStream_t stream) {
  tr_probe_kernel<<<1, 64, 0, stream>>>(static_cast<short*>(out), scheme);
}

}  // namespace colearn
