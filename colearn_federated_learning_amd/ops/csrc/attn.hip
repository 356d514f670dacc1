// Fused multi-head self-attention for gfx950 (BERT-base shapes:
// seq <= 128, dh = 64).  One block per (sample, head).
//
// Replaces the torch chain (score bmm + scale + fp32 softmax + dropout +
// context bmm + 3 reshape copies, fwd AND bwd — measured ~65% of BERT
// step time as small kernels, tools/gpu_debug profiling run): the whole
// head-attention runs out of LDS with MFMA tiles, reading q/k/v straight
// from the fused qkv projection layout [N, S, 3, H, dh] and writing ctx
// in [N, S, H*dh] — no transpose materialization anywhere.
//
// Dropout uses the same philox4x32-10 as the DP noise kernel, keyed on
// (seed, block, element): deterministic, layout-independent, and the
// backward REGENERATES the mask instead of storing it.  Softmax runs in
// fp32 registers (probs saved to global bf16 for the backward).
//
// fwd   : S = q k^T * scale ; P = softmax(S) ; C = dropout(P) V
// bwd 1 : dPd = dC V^T ; dP = dPd*mask ; dS = P (dP - rowsum(dP P))
//         dV = (P*mask)^T dC
// bwd 2 : dQ = dS K * scale ; dK = dS^T Q * scale
//
// MFMA fragment conventions match tile_gemm.h: A-frag row = lane&15,
// k = (lane>>4)*8 + j; C-frag row = (lane>>4)*4 + j, col = lane&15.

#include "api.h"
#include "tile_gemm.h"

namespace colearn {

namespace attn {

constexpr int S_MAX = 128;
constexpr int DH = 64;
// Unpadded tiles (LDS budget: bwd needs P(32K)+2x dh-tiles(16K) = 64 KB
// exactly) with slot-XOR swizzles; residual 2-way read conflicts are
// accepted (128/256-B pitches keep rows on the same banks).
constexpr int TDH_SH = DH;     // 128-B rows (q/k/v tiles), 16 KB each
constexpr int TSQ_SH = S_MAX;  // 256-B rows (P / dS tiles), 32 KB

DEV_INLINE int tdh(int row, int slot) {  // slot = 16-B group of 8 elems
  return row * (TDH_SH * 2) +
         ((slot ^ ((row + (row >> 3)) & 7)) * 16);
}
DEV_INLINE int tsq(int row, int slot) {
  return row * (TSQ_SH * 2) + ((slot ^ (row & 15)) * 16);
}

DEV_INLINE short8v_t frag_dh(const short* t, int row_base) {
  const int lane = threadIdx.x % kWave;
  return *reinterpret_cast<const short8v_t*>(
      reinterpret_cast<const char*>(t) +
      tdh(row_base + (lane & 15), lane >> 4));
}
DEV_INLINE short8v_t frag_dh_k(const short* t, int row_base, int k0) {
  const int lane = threadIdx.x % kWave;
  return *reinterpret_cast<const short8v_t*>(
      reinterpret_cast<const char*>(t) +
      tdh(row_base + (lane & 15), (k0 >> 3) + (lane >> 4)));
}
DEV_INLINE short8v_t frag_sq(const short* t, int row_base, int k0) {
  const int lane = threadIdx.x % kWave;
  return *reinterpret_cast<const short8v_t*>(
      reinterpret_cast<const char*>(t) +
      tsq(row_base + (lane & 15), (k0 >> 3) + (lane >> 4)));
}
// transposed A-frag from a [S][S] tile: row r of the TRANSPOSE = column r
DEV_INLINE short8v_t frag_sq_T(const short* t, int row_base, int k0) {
  const int lane = threadIdx.x % kWave;
  const int col = row_base + (lane & 15);      // column of the tile
  const int kq = k0 + (lane >> 4) * 8;         // tile row
  short8v_t v;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int row = kq + j;
    v[j] = *reinterpret_cast<const short*>(
        reinterpret_cast<const char*>(t) + tsq(row, col >> 3) +
        (col & 7) * 2);
  }
  return v;
}

// philox (shared impl pattern with dp_topk.hip): 4 uniforms per counter
DEV_INLINE void philox4(unsigned long long seed, unsigned long long ctr,
                        unsigned out[4]) {
  unsigned c0 = static_cast<unsigned>(ctr), c1 = static_cast<unsigned>(ctr >> 32);
  unsigned c2 = 0, c3 = 0;
  unsigned k0 = static_cast<unsigned>(seed),
           k1 = static_cast<unsigned>(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    const unsigned long long p0 = 0xD2511F53ull * c0;
    const unsigned long long p1 = 0xCD9E8D57ull * c2;
    const unsigned hi0 = static_cast<unsigned>(p0 >> 32),
                   lo0 = static_cast<unsigned>(p0);
    const unsigned hi1 = static_cast<unsigned>(p1 >> 32),
                   lo1 = static_cast<unsigned>(p1);
    const unsigned n0 = hi1 ^ c1 ^ k0, n2 = hi0 ^ c3 ^ k1;
    c0 = n0;
    c1 = lo1;
    c2 = n2;
    c3 = lo0;
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  out[0] = c0;
  out[1] = c1;
  out[2] = c2;
  out[3] = c3;
}

// keep-scale for element (i, j) of this block's P: 0 or 1/(1-p)
DEV_INLINE float drop_scale(unsigned long long seed, int64_t block,
                            int i, int j, float p, float inv_keep) {
  if (p <= 0.f) return 1.f;
  const int64_t e = (block * S_MAX + i) * S_MAX + j;
  unsigned r[4];
  philox4(seed, static_cast<unsigned long long>(e >> 2), r);
  const float u = (r[e & 3] >> 8) * (1.f / 16777216.f);
  return u >= p ? inv_keep : 0.f;
}

// stage a [rows][DH] bf16 tile from global (row stride rstride elems)
DEV_INLINE void stage_dh(const short* __restrict__ src, int64_t rstride,
                         int rows, short* dst) {
  // 256 threads: thread t covers (row = t>>3, slot = t&7), rows += 32
  const int r0 = threadIdx.x >> 3;
  const int slot = threadIdx.x & 7;
  for (int r = r0; r < rows; r += 32) {
    const short8v_t v = *reinterpret_cast<const short8v_t*>(
        src + static_cast<int64_t>(r) * rstride + slot * 8);
    *reinterpret_cast<short8v_t*>(
        reinterpret_cast<char*>(dst) + tdh(r, slot)) = v;
  }
}

// stage a [rows][S] bf16 tile from global (contiguous rows)
DEV_INLINE void stage_sq(const short* __restrict__ src, int rows, int S,
                         short* dst) {
  const int spr = S / 8;  // slots per row (16 at S=128)
  for (int e = threadIdx.x; e < rows * spr; e += 256) {
    const int r = e / spr, slot = e % spr;
    const short8v_t v = *reinterpret_cast<const short8v_t*>(
        src + static_cast<int64_t>(r) * S + slot * 8);
    *reinterpret_cast<short8v_t*>(
        reinterpret_cast<char*>(dst) + tsq(r, slot)) = v;
  }
}

}  // namespace attn

using namespace attn;

// ---------------------------------------------------------------------------
// forward: grid (N*H); qkv [N, S, 3, H, dh]; ctx [N, S, H*dh]; Psave
// [N*H, S, S] bf16 (softmax probs, pre-dropout)
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const short* __restrict__ qkv, short* __restrict__ ctx,
    short* __restrict__ Psave, unsigned long long* __restrict__ maskout,
    int N, int S, int H, float scale, float p,
    unsigned long long seed) {
  // 48 KB total; after the score phase the P tile OVERLAYS q+k (their
  // last read is fenced by the barrier below)
  __shared__ short smem[3 * S_MAX * TDH_SH];
  short* q_t = smem;
  short* k_t = smem + S_MAX * TDH_SH;
  short* v_t = smem + 2 * S_MAX * TDH_SH;
  short* p_t = smem;  // overlay (32 KB over q+k)
  const int64_t blk = blockIdx.x;
  const int n = static_cast<int>(blk) / H;
  const int h = static_cast<int>(blk) % H;
  const int64_t row_stride = static_cast<int64_t>(3) * H * DH;
  const short* base = qkv + (static_cast<int64_t>(n) * S) * row_stride +
                      static_cast<int64_t>(h) * DH;
  stage_dh(base, row_stride, S, q_t);                 // q
  stage_dh(base + static_cast<int64_t>(H) * DH, row_stride, S, k_t);
  stage_dh(base + static_cast<int64_t>(2) * H * DH, row_stride, S, v_t);
  __syncthreads();

  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int m0 = wid * 32;  // query rows of this wave
  // S-scores: [32 x S] per wave, frags acc[m][n]
  f32x4_t acc[2][8];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 8; ++n2) acc[m][n2] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int k0 = 0; k0 < DH; k0 += 32) {
    short8v_t aq[2], bk[8];
#pragma unroll
    for (int m = 0; m < 2; ++m) aq[m] = frag_dh_k(q_t, m0 + m * 16, k0);
#pragma unroll
    for (int n2 = 0; n2 < 8; ++n2)
      bk[n2] = frag_dh_k(k_t, n2 * 16, k0);
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n2 = 0; n2 < 8; ++n2)
        acc[m][n2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            aq[m], bk[n2], acc[m][n2], 0, 0, 0);
  }
  __syncthreads();  // q/k reads done on ALL waves: p_t may overlay them
  // softmax over each row (cols distributed: lane&15 + 16*n2); rows
  // rg*4+j with rg = lane>>4
  const float inv_keep = p > 0.f ? 1.f / (1.f - p) : 1.f;
  const int rg = lane >> 4;
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float mx = -1e30f;
#pragma unroll
      for (int n2 = 0; n2 < 8; ++n2) mx = fmaxf(mx, acc[m][n2][j]);
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off));
      float sum = 0.f;
#pragma unroll
      for (int n2 = 0; n2 < 8; ++n2) {
        acc[m][n2][j] = __expf((acc[m][n2][j] - mx) * scale);
        sum += acc[m][n2][j];
      }
      for (int off = 1; off < 16; off <<= 1)
        sum += __shfl_xor(sum, off);
      const float inv = 1.f / sum;
#pragma unroll
      for (int n2 = 0; n2 < 8; ++n2) acc[m][n2][j] *= inv;
    }
  }
  // save probs (bf16) to global + stage dropped probs into LDS; the
  // dropout decisions are ALSO saved as one u64 bitfield per thread
  // (fragment-ordered) so the backward never re-runs philox
  short* Pg = Psave + blk * S * S;
  unsigned long long mbits = 0;
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 8; ++n2)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int row = m0 + m * 16 + rg * 4 + j;
        const int col = n2 * 16 + (lane & 15);
        const float pr = acc[m][n2][j];
        Pg[row * S + col] = f2bits(pr);
        const float ds = drop_scale(seed, blk, row, col, p, inv_keep);
        if (ds > 0.f)
          mbits |= 1ull << ((m * 8 + n2) * 4 + j);
        *reinterpret_cast<short*>(
            reinterpret_cast<char*>(p_t) + tsq(row, col >> 3) +
            (col & 7) * 2) = f2bits(pr * ds);
      }
  maskout[blk * 256 + threadIdx.x] = mbits;
  __syncthreads();
  // C = Pd @ V : [32 x DH] per wave
  f32x4_t cacc[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 4; ++n2) cacc[m][n2] = {0.f, 0.f, 0.f, 0.f};
  for (int k0 = 0; k0 < S; k0 += 32) {
    short8v_t ap[2], bv[4];
#pragma unroll
    for (int m = 0; m < 2; ++m) ap[m] = frag_sq(p_t, m0 + m * 16, k0);
    // B operand: V^T? no — C[i][d] = sum_j Pd[i][j] V[j][d]: B-frag rows
    // = output cols d, k = j: V is [S][DH] (k=j strided) -> transposed
    // read of v_t
    {
      const int lane2 = threadIdx.x % kWave;
      const int d0 = lane2 & 15;
      const int kq = k0 + (lane2 >> 4) * 8;
#pragma unroll
      for (int n2 = 0; n2 < 4; ++n2) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int vrow = kq + j;
          const int vcol = n2 * 16 + d0;
          bv[n2][j] = *reinterpret_cast<const short*>(
              reinterpret_cast<const char*>(v_t) + tdh(vrow, vcol >> 3) +
              (vcol & 7) * 2);
        }
      }
    }
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n2 = 0; n2 < 4; ++n2)
        cacc[m][n2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            ap[m], bv[n2], cacc[m][n2], 0, 0, 0);
  }
  // write ctx [N, S, H*dh]
  short* cg = ctx + (static_cast<int64_t>(n) * S) * H * DH +
              static_cast<int64_t>(h) * DH;
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 4; ++n2)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int row = m0 + m * 16 + rg * 4 + j;
        const int col = n2 * 16 + (lane & 15);
        cg[static_cast<int64_t>(row) * H * DH + col] =
            f2bits(cacc[m][n2][j]);
      }
}

// ---------------------------------------------------------------------------
// backward 1: dV and dS.  grid (N*H).
//   dPd = dC V^T ; dP = dPd * mask ; dS = P (dP - rowsum(dP P)) * scale
//   dV = (P*mask)^T dC
// dS is written to global (bf16) for kernel 2.
__global__ __launch_bounds__(256) void attn_bwd1_kernel(
    const short* __restrict__ dctx, const short* __restrict__ qkv,
    const short* __restrict__ Psave,
    const unsigned long long* __restrict__ maskin, short* __restrict__ dS_g,
    short* __restrict__ dqkv, int N, int S, int H, float scale, float p,
    unsigned long long seed) {
  __shared__ short smem[S_MAX * TSQ_SH + 2 * S_MAX * TDH_SH];  // 64 KB
  short* p_t = smem;
  short* dc_t = smem + S_MAX * TSQ_SH;
  short* v_t = dc_t + S_MAX * TDH_SH;
  const int64_t blk = blockIdx.x;
  const int n = static_cast<int>(blk) / H;
  const int h = static_cast<int>(blk) % H;
  const int64_t qkv_stride = static_cast<int64_t>(3) * H * DH;
  const short* vsrc = qkv + (static_cast<int64_t>(n) * S) * qkv_stride +
                      (static_cast<int64_t>(2) * H + h) * DH;
  const short* dcsrc = dctx + (static_cast<int64_t>(n) * S) * H * DH +
                       static_cast<int64_t>(h) * DH;
  stage_dh(dcsrc, static_cast<int64_t>(H) * DH, S, dc_t);
  stage_dh(vsrc, qkv_stride, S, v_t);
  stage_sq(Psave + blk * S * S, S, S, p_t);
  __syncthreads();

  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int m0 = wid * 32;
  const int rg = lane >> 4;
  const float inv_keep = p > 0.f ? 1.f / (1.f - p) : 1.f;
  const unsigned long long mbits = maskin[blk * 256 + threadIdx.x];

  // dPd = dC @ V^T : [32 x S] per wave (A = dC rows, B = V rows)
  f32x4_t acc[2][8];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 8; ++n2) acc[m][n2] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int k0 = 0; k0 < DH; k0 += 32) {
    short8v_t a[2], b[8];
#pragma unroll
    for (int m = 0; m < 2; ++m) a[m] = frag_dh_k(dc_t, m0 + m * 16, k0);
#pragma unroll
    for (int n2 = 0; n2 < 8; ++n2)
      b[n2] = frag_dh_k(v_t, n2 * 16, k0);
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n2 = 0; n2 < 8; ++n2)
        acc[m][n2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[m], b[n2], acc[m][n2], 0, 0, 0);
  }
  // dP = dPd * mask ; rowdot = sum_j dP*P ; dS = P*(dP - rowdot)*scale
  short* dsg = dS_g + blk * S * S;
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int row = m0 + m * 16 + rg * 4 + j;
      float pr[8], dp[8];
      float rd = 0.f;
#pragma unroll
      for (int n2 = 0; n2 < 8; ++n2) {
        const int col = n2 * 16 + (lane & 15);
        pr[n2] = bits2f(*reinterpret_cast<const short*>(
            reinterpret_cast<const char*>(p_t) + tsq(row, col >> 3) +
            (col & 7) * 2));
        const float ds =
            (mbits >> ((m * 8 + n2) * 4 + j)) & 1 ? inv_keep : 0.f;
        dp[n2] = acc[m][n2][j] * (p > 0.f ? ds : 1.f);
        rd += dp[n2] * pr[n2];
      }
      for (int off = 1; off < 16; off <<= 1) rd += __shfl_xor(rd, off);
#pragma unroll
      for (int n2 = 0; n2 < 8; ++n2) {
        const int col = n2 * 16 + (lane & 15);
        dsg[row * S + col] =
            f2bits(pr[n2] * (dp[n2] - rd) * scale);
      }
    }
  }
  __syncthreads();
  // overwrite p_t in place with Pd = P * mask (for dV) — each thread
  // scales exactly the elements its mask bitfield covers (fragment
  // order), so no philox runs in the backward at all
  if (p > 0.f) {
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n2 = 0; n2 < 8; ++n2)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int row = m0 + m * 16 + rg * 4 + j;
          const int col = n2 * 16 + (lane & 15);
          const float ds =
              (mbits >> ((m * 8 + n2) * 4 + j)) & 1 ? inv_keep : 0.f;
          short* cell = reinterpret_cast<short*>(
              reinterpret_cast<char*>(p_t) + tsq(row, col >> 3) +
              (col & 7) * 2);
          *cell = f2bits(bits2f(*cell) * ds);
        }
  }
  __syncthreads();
  // dV = Pd^T @ dC : [32 key rows x DH] per wave; A = Pd^T (transposed
  // read of p_t), B = dC^T-frag (transposed read of dc_t, like fwd C)
  f32x4_t vacc[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 4; ++n2) vacc[m][n2] = {0.f, 0.f, 0.f, 0.f};
  for (int k0 = 0; k0 < S; k0 += 32) {
    short8v_t a[2], b[4];
#pragma unroll
    for (int m = 0; m < 2; ++m)
      a[m] = frag_sq_T(p_t, m0 + m * 16, k0);
    {
      const int d0 = lane & 15;
      const int kq = k0 + (lane >> 4) * 8;
#pragma unroll
      for (int n2 = 0; n2 < 4; ++n2)
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int col = n2 * 16 + d0;
          b[n2][j] = *reinterpret_cast<const short*>(
              reinterpret_cast<const char*>(dc_t) + tdh(kq + j, col >> 3) +
              (col & 7) * 2);
        }
    }
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n2 = 0; n2 < 4; ++n2)
        vacc[m][n2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[m], b[n2], vacc[m][n2], 0, 0, 0);
  }
  short* dvg = dqkv + (static_cast<int64_t>(n) * S) * qkv_stride +
               (static_cast<int64_t>(2) * H + h) * DH;
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 4; ++n2)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int row = m0 + m * 16 + rg * 4 + j;
        const int col = n2 * 16 + (lane & 15);
        dvg[static_cast<int64_t>(row) * qkv_stride + col] =
            f2bits(vacc[m][n2][j]);
      }
}

// ---------------------------------------------------------------------------
// backward 2: dQ = dS @ K * 1 ; dK = dS^T @ Q * 1  (scale already folded
// into dS by kernel 1).  grid (N*H).
__global__ __launch_bounds__(256) void attn_bwd2_kernel(
    const short* __restrict__ dS_g, const short* __restrict__ qkv,
    short* __restrict__ dqkv, int N, int S, int H) {
  __shared__ short smem[S_MAX * TSQ_SH + 2 * S_MAX * TDH_SH];  // 64 KB
  short* ds_t = smem;
  short* q_t = smem + S_MAX * TSQ_SH;
  short* k_t = q_t + S_MAX * TDH_SH;
  const int64_t blk = blockIdx.x;
  const int n = static_cast<int>(blk) / H;
  const int h = static_cast<int>(blk) % H;
  const int64_t qkv_stride = static_cast<int64_t>(3) * H * DH;
  const short* base = qkv + (static_cast<int64_t>(n) * S) * qkv_stride +
                      static_cast<int64_t>(h) * DH;
  stage_dh(base, qkv_stride, S, q_t);
  stage_dh(base + static_cast<int64_t>(H) * DH, qkv_stride, S, k_t);
  stage_sq(dS_g + blk * S * S, S, S, ds_t);
  __syncthreads();

  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int m0 = wid * 32;
  const int rg = lane >> 4;

  // dQ = dS @ K : [32 x DH]; B-frag = transposed read of k_t (cols d)
  f32x4_t qacc[2][4], kacc[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 4; ++n2) {
      qacc[m][n2] = {0.f, 0.f, 0.f, 0.f};
      kacc[m][n2] = {0.f, 0.f, 0.f, 0.f};
    }
  for (int k0 = 0; k0 < S; k0 += 32) {
    short8v_t a[2], at[2], bK[4], bQ[4];
#pragma unroll
    for (int m = 0; m < 2; ++m) {
      a[m] = frag_sq(ds_t, m0 + m * 16, k0);     // dS rows
      at[m] = frag_sq_T(ds_t, m0 + m * 16, k0);  // dS^T rows
    }
    {
      const int d0 = lane & 15;
      const int kq = k0 + (lane >> 4) * 8;
#pragma unroll
      for (int n2 = 0; n2 < 4; ++n2)
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int col = n2 * 16 + d0;
          bK[n2][j] = *reinterpret_cast<const short*>(
              reinterpret_cast<const char*>(k_t) + tdh(kq + j, col >> 3) +
              (col & 7) * 2);
          bQ[n2][j] = *reinterpret_cast<const short*>(
              reinterpret_cast<const char*>(q_t) + tdh(kq + j, col >> 3) +
              (col & 7) * 2);
        }
    }
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n2 = 0; n2 < 4; ++n2) {
        qacc[m][n2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[m], bK[n2], qacc[m][n2], 0, 0, 0);
        kacc[m][n2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            at[m], bQ[n2], kacc[m][n2], 0, 0, 0);
      }
  }
  short* dqg = dqkv + (static_cast<int64_t>(n) * S) * qkv_stride +
               static_cast<int64_t>(h) * DH;
  short* dkg = dqg + static_cast<int64_t>(H) * DH;
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n2 = 0; n2 < 4; ++n2)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int row = m0 + m * 16 + rg * 4 + j;
        const int col = n2 * 16 + (lane & 15);
        dqg[static_cast<int64_t>(row) * qkv_stride + col] =
            f2bits(qacc[m][n2][j]);
        dkg[static_cast<int64_t>(row) * qkv_stride + col] =
            f2bits(kacc[m][n2][j]);
      }
}

// ---- launchers ------------------------------------------------------------
void launch_attn_fwd(const void* qkv, void* ctx, void* Psave,
                     unsigned long long* mask, int N, int S, int H,
                     float scale, float p, unsigned long long seed,
                     hipStream_t stream) {
  attn_fwd_kernel<<<N * H, 256, 0, stream>>>(
      static_cast<const short*>(qkv), static_cast<short*>(ctx),
      static_cast<short*>(Psave), mask, N, S, H, scale, p, seed);
}

void launch_attn_bwd(const void* dctx, const void* qkv, const void* Psave,
                     const unsigned long long* mask, void* dS, void* dqkv,
                     int N, int S, int H, float scale, float p,
                     unsigned long long seed, hipStream_t stream) {
  attn_bwd1_kernel<<<N * H, 256, 0, stream>>>(
      static_cast<const short*>(dctx), static_cast<const short*>(qkv),
      static_cast<const short*>(Psave), mask, static_cast<short*>(dS),
      static_cast<short*>(dqkv), N, S, H, scale, p, seed);
  attn_bwd2_kernel<<<N * H, 256, 0, stream>>>(
      static_cast<const short*>(dS), static_cast<const short*>(qkv),
      static_cast<short*>(dqkv), N, S, H);
}

}  // namespace colearn
