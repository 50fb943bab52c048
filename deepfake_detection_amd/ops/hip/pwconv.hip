// Pointwise (1x1) convolution on MFMA matrix cores, NHWC, gfx950.
//
// Forward:    y[M, N] = x[M, K] @ w[N, K]^T   (M = B*H*W, K = C_in, N = C_out;
//             the torch conv weight (C_out, C_in, 1, 1) is ALREADY the [N, K]
//             B-transposed layout the kernel wants — no repacking).
// Bwd-data:   dx[M, K] = dy[M, N] @ w[N, K] — the SAME forward kernel with w
//             transposed once on the host (tiny).
// Bwd-weight: dW[N, K] = dy^T[N, M] @ x[M, K] — a tall reduction over M.
//             Split-M two-stage: blocks own (n-tile, k-tile, m-chunk), stage
//             dy/x tiles TRANSPOSED through LDS (contraction dim M is the
//             row-major stride-C dim, so MFMA fragments need m-contiguous
//             reads), write fp32 partials per chunk, then a reduce kernel
//             sums chunks into the bf16 grad. No atomics.
//
// Optional forward stats epilogue (template STATS): the kernel accumulates
// per-channel sum/sumsq of the OUTPUT tile into 64 bucketed fp32 partial
// buffers ([64, 2, N], bucket = blockIdx.x & 63 so contention per address is
// grid.x/64 spread over the kernel) — the following BatchNorm consumes them
// via bn_act_fwd(..., stats) and skips its own full read of y
// (the single biggest measured cost in r01 profiles).
//
// Fragment mappings (cdna4 16x16x32 bf16):
//   A/B: row(col) = lane & 15, k = (lane >> 4) * 8 + i   (8 bf16 / lane)
//   C/D: col = lane & 15, row = (lane >> 4) * 4 + reg    (4 fp32 / lane)
//
// Replaces MIOpen igemm for the reference's create_conv2d 1x1 call sites
// (reference dfd/timm/models/efficientnet_blocks.py:277,299; dispatch
// ops/pw_dispatch.py).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int kStatsBuckets = 64;

// ---------------------------------------------------------------------------
// forward GEMM: double-buffered LDS pipeline, BK=64 (two MFMA k-steps per
// stage), templated block/wave tiling:
//   big    128(M) x 128(N), 4 waves in 2x2, wave tile 64x64 (4x4 frags)
//   skinny 128(M) x  32(N), 4 waves in 4x1, wave tile 32x32 (2x2 frags)
//     (for the MBConv pw-linear projections: N = 24..48 would waste 3/4 of
//      a 128-wide N tile)
// Per iteration each wave issues its next-tile global loads into registers,
// runs the MFMAs on the current LDS buffer, then writes the registers into
// the other buffer — one __syncthreads per K step.
// ---------------------------------------------------------------------------
// KSTEP: staged K per iteration (one or two MFMA k-steps); 32 for the
// low-K expansion convs (e.g. 24->144) where a 64-wide stage is half zeros.
template <int R, int KS>
constexpr int pw_units() { return (R * KS / 8 + 255) / 256; }

template <int ROWS, int KSTEP>
DFD_DEV void pw_load_tile(const __hip_bfloat16* __restrict__ src, long long row0,
                          long long row_end, int k0, int K, int ld, int tid,
                          bf16x8 (&v)[pw_units<ROWS, KSTEP>()]) {
#pragma unroll
  for (int u = 0; u < pw_units<ROWS, KSTEP>(); ++u) {
    const int idx = tid + u * 256;
    if (idx >= ROWS * KSTEP / 8) break;
    const int row = idx / (KSTEP / 8);
    const int c = (idx % (KSTEP / 8)) * 8;
    const long long gr = row0 + row;
    v[u] = bf16x8{};
    if (gr < row_end) {
      // vec path needs 16-B alignment: row base gr*ld is only aligned when
      // ld (=K) is a multiple of 8 elements
      if ((ld & 7) == 0 && k0 + c + 7 < K) {
        v[u] = *reinterpret_cast<const bf16x8*>(src + gr * (long long)ld + k0 + c);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int kk = k0 + c + e;
          reinterpret_cast<__bf16*>(&v[u])[e] =
              kk < K ? *reinterpret_cast<const __bf16*>(src + gr * (long long)ld + kk)
                     : (__bf16)0.f;
        }
      }
    }
  }
}

template <int ROWS, int KSTEP>
DFD_DEV void pw_store_tile(__bf16* lds, int tid,
                           const bf16x8 (&v)[pw_units<ROWS, KSTEP>()]) {
#pragma unroll
  for (int u = 0; u < pw_units<ROWS, KSTEP>(); ++u) {
    const int idx = tid + u * 256;
    if (idx >= ROWS * KSTEP / 8) break;
    *reinterpret_cast<bf16x8*>(
        &lds[(idx / (KSTEP / 8)) * (KSTEP + 8) + (idx % (KSTEP / 8)) * 8]) = v[u];
  }
}

// NBUF: LDS pipeline depth — 1 for single-iteration shapes (K <= BK), where
// the 2nd buffer would only halve occupancy, 2 (double-buffered) otherwise.
template <int BM, int BN, int WROWS, int WCOLS, int NBUF, int KSTEP, bool STATS>
__global__ __launch_bounds__(256) void pw_gemm_bf16_kernel(
    const __hip_bfloat16* __restrict__ x,  // [M, K] row-major
    const __hip_bfloat16* __restrict__ w,  // [N, K] row-major
    __hip_bfloat16* __restrict__ y,        // [M, N] row-major
    float* __restrict__ stats,             // [kStatsBuckets, 2, N] or null
    long long M, int N, int K) {
  constexpr int WTM = BM / WROWS;            // wave tile M
  constexpr int WTN = BN / WCOLS;            // wave tile N
  constexpr int FI = WTM / 16;               // row fragments per wave
  constexpr int FJ = WTN / 16;               // col fragments per wave
  constexpr int LDK = KSTEP + 8;             // padded LDS row stride
  constexpr int STAGE = NBUF * (BM + BN) * LDK;
  constexpr int CTILE = 4 * WTM * WTN;       // writeback staging
  constexpr int SMEM = STAGE > CTILE ? STAGE : CTILE;
  __shared__ __bf16 smem[SMEM];
  auto a_buf = [&](int b) { return &smem[b * BM * LDK]; };
  auto b_buf = [&](int b) { return &smem[NBUF * BM * LDK + b * BN * LDK]; };

  const long long m0 = (long long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wid = tid / kWave;
  const int wm = (wid % WROWS) * WTM;
  const int wn = (wid / WROWS) * WTN;

  f32x4 acc[FI][FJ];
#pragma unroll
  for (int i = 0; i < FI; ++i)
#pragma unroll
    for (int j = 0; j < FJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int lrow = lane & 15;      // fragment row/col
  const int lk = (lane >> 4) * 8;  // fragment k offset within a 32-k step

  bf16x8 av[pw_units<BM, KSTEP>()], bv[pw_units<BN, KSTEP>()];
  pw_load_tile<BM, KSTEP>(x, m0, M, 0, K, K, tid, av);
  pw_load_tile<BN, KSTEP>(w, n0, N, 0, K, K, tid, bv);
  pw_store_tile<BM, KSTEP>(a_buf(0), tid, av);
  pw_store_tile<BN, KSTEP>(b_buf(0), tid, bv);
  __syncthreads();

  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += KSTEP, cur ^= (NBUF - 1)) {
    const bool has_next = k0 + KSTEP < K;
    if (has_next) {
      pw_load_tile<BM, KSTEP>(x, m0, M, k0 + KSTEP, K, K, tid, av);
      pw_load_tile<BN, KSTEP>(w, n0, N, k0 + KSTEP, K, K, tid, bv);
    }
#pragma unroll
    for (int s = 0; s < KSTEP / 32; ++s) {
      bf16x8 afrag[FI], bfrag[FJ];
#pragma unroll
      for (int i = 0; i < FI; ++i)
        afrag[i] = *reinterpret_cast<const bf16x8*>(
            &a_buf(cur)[(wm + i * 16 + lrow) * LDK + s * 32 + lk]);
#pragma unroll
      for (int j = 0; j < FJ; ++j)
        bfrag[j] = *reinterpret_cast<const bf16x8*>(
            &b_buf(cur)[(wn + j * 16 + lrow) * LDK + s * 32 + lk]);
#pragma unroll
      for (int i = 0; i < FI; ++i)
#pragma unroll
        for (int j = 0; j < FJ; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    if (has_next) {
      pw_store_tile<BM, KSTEP>(a_buf(cur ^ 1), tid, av);
      pw_store_tile<BN, KSTEP>(b_buf(cur ^ 1), tid, bv);
    }
    __syncthreads();
  }

  // ---- epilogue ----
  // C/D fragment: col = lane&15, row = (lane>>4)*4 + reg.
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;

  if (STATS) {
    // per-channel sum/sumsq of the ROUNDED outputs (matches a separate pass
    // over the stored bf16 y)
    float ssum[FJ], sq[FJ];
#pragma unroll
    for (int j = 0; j < FJ; ++j) { ssum[j] = 0.f; sq[j] = 0.f; }
#pragma unroll
    for (int i = 0; i < FI; ++i)
#pragma unroll
      for (int j = 0; j < FJ; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const long long gm = m0 + wm + i * 16 + crow0 + r;
          const int gn = n0 + wn + j * 16 + ccol;
          if (gm < M && gn < N) {
            const float f = __bfloat162float(__float2bfloat16(acc[i][j][r]));
            ssum[j] += f;
            sq[j] += f * f;
          }
        }
    // rows 16/32/48 fold onto lanes 0-15 (same output column)
#pragma unroll
    for (int j = 0; j < FJ; ++j) {
      ssum[j] += __shfl_down(ssum[j], 32);
      ssum[j] += __shfl_down(ssum[j], 16);
      sq[j] += __shfl_down(sq[j], 32);
      sq[j] += __shfl_down(sq[j], 16);
    }
    __syncthreads();  // LDS rewritten as fp32 scratch below
    float* sscr = reinterpret_cast<float*>(&smem[0]);  // [4 waves][WTN]
    float* qscr = sscr + 4 * WTN;
    if (lane < 16) {
#pragma unroll
      for (int j = 0; j < FJ; ++j) {
        sscr[wid * WTN + j * 16 + lane] = ssum[j];
        qscr[wid * WTN + j * 16 + lane] = sq[j];
      }
    }
    __syncthreads();
    if (tid < BN) {
      const int col = tid;                 // n-column within the block tile
      const int q = col / WTN;             // column-group index
      const int sub = col % WTN;
      const int gn = n0 + col;
      if (gn < N) {
        float sv = 0.f, qv = 0.f;
#pragma unroll
        for (int rr = 0; rr < WROWS; ++rr) {
          sv += sscr[(q * WROWS + rr) * WTN + sub];
          qv += qscr[(q * WROWS + rr) * WTN + sub];
        }
        float* bucket = stats + (size_t)(blockIdx.x & (kStatsBuckets - 1)) * 2 * N;
        atomicAdd(bucket + gn, sv);
        atomicAdd(bucket + N + gn, qv);
      }
    }
    __syncthreads();  // before the writeback re-uses the LDS
  }

  // Vectorized writeback: per-lane scalar b16 stores are store-ISSUE-bound
  // (cdna_hip_programming.md T21), so each wave round-trips its tile through
  // LDS and issues 16-B row-major stores instead.
  {
    __bf16* ctile = &smem[0] + wid * (WTM * WTN);
#pragma unroll
    for (int i = 0; i < FI; ++i)
#pragma unroll
      for (int j = 0; j < FJ; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          ctile[(i * 16 + crow0 + r) * WTN + j * 16 + ccol] =
              (__bf16)__float2bfloat16(acc[i][j][r]);
    __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt: wave's own LDS stores landed
    constexpr int CU = WTM * WTN / 64 / 8;  // vec8 units per lane
    const bool n_aligned = (N & 7) == 0;
#pragma unroll
    for (int u = 0; u < CU; ++u) {
      const int unit = lane + u * 64;
      const int row = unit / (WTN / 8);
      const int c8 = (unit % (WTN / 8)) * 8;
      const long long gm = m0 + wm + row;
      if (gm >= M) continue;
      const int gn = n0 + wn + c8;
      if (n_aligned && gn + 8 <= N) {
        *reinterpret_cast<bf16x8*>(y + gm * N + gn) =
            *reinterpret_cast<const bf16x8*>(&ctile[row * WTN + c8]);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (gn + e < N)
            *reinterpret_cast<__bf16*>(y + gm * N + gn + e) = ctile[row * WTN + c8 + e];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// small-K expansion GEMM (K <= 32, 64 < N <= 192): the whole contraction fits
// one MFMA k-step, and the A/B fragment layouts are row-major vec8 per lane —
// so fragments load DIRECTLY from global memory. No LDS, no barriers; weight
// fragments load once per thread and m-tiles grid-stride. Stats accumulate in
// registers across all m-tiles (one bucket atomic set per block at the end).
// PMC (r02): the staged kernel ran this class at 34% of roofline, 64%
// wave-parked on its barriers.
// ---------------------------------------------------------------------------
template <int FJ, bool STATS>
__global__ __launch_bounds__(256) void pw_gemm_smallk_kernel(
    const __hip_bfloat16* __restrict__ x,  // [M, K] row-major
    const __hip_bfloat16* __restrict__ w,  // [N, K] row-major
    __hip_bfloat16* __restrict__ y,        // [M, N] row-major
    float* __restrict__ stats,             // [kStatsBuckets, 2, N] or null
    long long M, int N, int K) {
  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wid = tid / kWave;
  const int lrow = lane & 15;
  const int lk = (lane >> 4) * 8;
  const bool kin = lk < K;

  // B fragments once per thread (w is tiny and L1/L2-hot)
  bf16x8 bfrag[FJ];
#pragma unroll
  for (int j = 0; j < FJ; ++j) {
    const int gn = j * 16 + lrow;
    bfrag[j] = bf16x8{};
    if (gn < N && kin)
      bfrag[j] = *reinterpret_cast<const bf16x8*>(w + (long long)gn * K + lk);
  }

  float ssum[FJ], sq[FJ];
#pragma unroll
  for (int j = 0; j < FJ; ++j) { ssum[j] = 0.f; sq[j] = 0.f; }

  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  const long long mtiles = (M + 255) / 256;  // 4 waves x 64 rows
  for (long long mt = blockIdx.x; mt < mtiles; mt += gridDim.x) {
    const long long m0 = mt * 256 + wid * 64;
    f32x4 acc[4][FJ];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < FJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    bf16x8 afrag[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const long long gm = m0 + i * 16 + lrow;
      afrag[i] = bf16x8{};
      if (gm < M && kin)
        afrag[i] = *reinterpret_cast<const bf16x8*>(x + gm * K + lk);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < FJ; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[i], bfrag[j], acc[i][j], 0, 0, 0);

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int j = 0; j < FJ; ++j) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const long long gm = m0 + i * 16 + crow0 + r;
          const int gn = j * 16 + ccol;
          if (gm < M && gn < N) {
            const __hip_bfloat16 v = __float2bfloat16(acc[i][j][r]);
            y[gm * N + gn] = v;
            if (STATS) {
              const float f = __bfloat162float(v);
              ssum[j] += f;
              sq[j] += f * f;
            }
          }
        }
      }
    }
  }

  if (STATS) {
#pragma unroll
    for (int j = 0; j < FJ; ++j) {
      ssum[j] += __shfl_down(ssum[j], 32);
      ssum[j] += __shfl_down(ssum[j], 16);
      sq[j] += __shfl_down(sq[j], 32);
      sq[j] += __shfl_down(sq[j], 16);
    }
    if (lane < 16) {
      float* bucket =
          stats + (size_t)((blockIdx.x * 4 + wid) & (kStatsBuckets - 1)) * 2 * N;
#pragma unroll
      for (int j = 0; j < FJ; ++j) {
        const int gn = j * 16 + lane;
        if (gn < N) {
          atomicAdd(bucket + gn, ssum[j]);
          atomicAdd(bucket + N + gn, sq[j]);
        }
      }
    }
  }
}

constexpr int kPwBM = 128;
// ---------------------------------------------------------------------------
// bwd-weight: dW[N, K] = dy^T @ x, split over M into chunks of fp32 partials.
// block 256 = 4 waves; block tile 64(N) x 64(K); wave 32x32; TM=64 per stage.
// ---------------------------------------------------------------------------
constexpr int WT = 64;        // block tile along N and K
constexpr int TM = 64;        // m elements staged per iteration
constexpr int LDM = TM + 8;   // padded LDS row (16-B aligned stride: 144 B)

__global__ __launch_bounds__(256) void pw_wgrad_kernel(
    const __hip_bfloat16* __restrict__ dy,  // [M, N]
    const __hip_bfloat16* __restrict__ x,   // [M, K]
    float* __restrict__ part,               // [chunks, N, K]
    long long M, int N, int K, long long rows_per_chunk) {
  __shared__ __bf16 dyt[WT * LDM];  // [n][m] transposed
  __shared__ __bf16 xt[WT * LDM];   // [k][m] transposed

  const int ktiles = (K + WT - 1) / WT;
  const int n0 = (blockIdx.x / ktiles) * WT;
  const int k0 = (blockIdx.x % ktiles) * WT;
  const long long r0 = (long long)blockIdx.y * rows_per_chunk;
  const long long r1 = min(r0 + rows_per_chunk, M);

  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wid = tid / kWave;
  const int wn = (wid & 1) * 32;
  const int wk = (wid >> 1) * 32;
  const int lrow = lane & 15;
  const int lk = (lane >> 4) * 8;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // stage thread plan: m-local = tid & 63, column quarter = tid >> 6
  const int sm = tid & 63;
  const int sq = tid >> 6;  // 0..3 -> 16-column group

  for (long long m0 = r0; m0 < r1; m0 += TM) {
    const long long gm = m0 + sm;
    // dy tile: load vec8 along n, scatter-transpose into [n][m]
    {
      bf16x8 v[2];
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const int n = n0 + sq * 16 + h * 8;
        v[h] = bf16x8{};
        if (gm < r1 && (N & 7) == 0 && n + 7 < N) {
          v[h] = *reinterpret_cast<const bf16x8*>(dy + gm * N + n);
        } else if (gm < r1) {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            reinterpret_cast<__bf16*>(&v[h])[e] =
                (n + e) < N ? *reinterpret_cast<const __bf16*>(dy + gm * N + n + e)
                            : (__bf16)0.f;
        }
      }
#pragma unroll
      for (int h = 0; h < 2; ++h)
#pragma unroll
        for (int e = 0; e < 8; ++e)
          dyt[(sq * 16 + h * 8 + e) * LDM + sm] = reinterpret_cast<__bf16*>(&v[h])[e];
    }
    // x tile: same, into [k][m]
    {
      bf16x8 v[2];
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const int k = k0 + sq * 16 + h * 8;
        v[h] = bf16x8{};
        if (gm < r1 && (K & 7) == 0 && k + 7 < K) {
          v[h] = *reinterpret_cast<const bf16x8*>(x + gm * K + k);
        } else if (gm < r1) {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            reinterpret_cast<__bf16*>(&v[h])[e] =
                (k + e) < K ? *reinterpret_cast<const __bf16*>(x + gm * K + k + e)
                            : (__bf16)0.f;
        }
      }
#pragma unroll
      for (int h = 0; h < 2; ++h)
#pragma unroll
        for (int e = 0; e < 8; ++e)
          xt[(sq * 16 + h * 8 + e) * LDM + sm] = reinterpret_cast<__bf16*>(&v[h])[e];
    }
    __syncthreads();

    // 2 MFMA contraction steps of 32 m each
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      bf16x8 afrag[2], bfrag[2];
#pragma unroll
      for (int i = 0; i < 2; ++i)
        afrag[i] = *reinterpret_cast<const bf16x8*>(
            &dyt[(wn + i * 16 + lrow) * LDM + s * 32 + lk]);
#pragma unroll
      for (int j = 0; j < 2; ++j)
        bfrag[j] = *reinterpret_cast<const bf16x8*>(
            &xt[(wk + j * 16 + lrow) * LDM + s * 32 + lk]);
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // partial writeback (fp32, per chunk; reduce kernel sums the chunk axis)
  float* out = part + (size_t)blockIdx.y * N * K;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gn = n0 + wn + i * 16 + crow0 + r;
        const int gk = k0 + wk + j * 16 + ccol;
        if (gn < N && gk < K) out[(size_t)gn * K + gk] = acc[i][j][r];
      }
    }
  }
}

__global__ void pw_wgrad_reduce_kernel(const float* __restrict__ part,
                                       __hip_bfloat16* __restrict__ dw,
                                       long long nk, int chunks) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nk) return;
  float s = 0.f;
  for (int c = 0; c < chunks; ++c) s += part[(size_t)c * nk + i];
  dw[i] = __float2bfloat16(s);
}

}  // namespace

// x: (B, C_in, H, W) channels_last; w: (C_out, C_in, 1, 1). Returns NHWC y
// (and fills `stats` [64, 2, C_out] fp32 zero-initialized, if provided).
at::Tensor pw_conv2d_fwd_mfma(at::Tensor x, at::Tensor w,
                              c10::optional<at::Tensor> stats_opt) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda(), "pwconv: CUDA tensors required");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kBFloat16,
              "pwconv: bf16 only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "pwconv: channels_last input required");
  TORCH_CHECK(w.size(2) == 1 && w.size(3) == 1, "pwconv: 1x1 kernels only");
  const long long M = (long long)x.size(0) * x.size(2) * x.size(3);
  const int K = (int)x.size(1);
  const int N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K, "pwconv: channel mismatch");
  auto wc = w.contiguous();  // [N, K] row-major
  auto y = at::empty({x.size(0), (long long)N, x.size(2), x.size(3)},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto stream = at::hip::getCurrentHIPStream().stream();
  float* stats_p = nullptr;
  if (stats_opt.has_value()) {
    at::Tensor stats = *stats_opt;
    TORCH_CHECK(stats.is_cuda() && stats.scalar_type() == at::kFloat &&
                    stats.numel() == (long long)kStatsBuckets * 2 * N &&
                    stats.is_contiguous(),
                "pwconv: stats must be fp32 [64, 2, C_out] contiguous");
    stats_p = stats.data_ptr<float>();
  }
  const auto* xp = (const __hip_bfloat16*)x.data_ptr();
  const auto* wp = (const __hip_bfloat16*)wc.data_ptr();
  auto* yp = (__hip_bfloat16*)y.data_ptr();
  // N-tile config: smallest width that doesn't add extra column tiles
  // (every extra column tile re-reads the whole x) — 32 for the pw-linear
  // projections (N<=32), 64 for N<=64, 128 otherwise.
#define PW_KERNEL(BN_, WR_, WC_, NBUF_, KS_)                                  \
  do {                                                                        \
    dim3 grid((unsigned)((M + kPwBM - 1) / kPwBM), (N + (BN_)-1) / (BN_));    \
    if (stats_p)                                                              \
      pw_gemm_bf16_kernel<128, BN_, WR_, WC_, NBUF_, KS_, true>               \
          <<<grid, 256, 0, stream>>>(xp, wp, yp, stats_p, M, N, K);           \
    else                                                                      \
      pw_gemm_bf16_kernel<128, BN_, WR_, WC_, NBUF_, KS_, false>              \
          <<<grid, 256, 0, stream>>>(xp, wp, yp, nullptr, M, N, K);           \
  } while (0)

#define PW_LAUNCH(BN_, WR_, WC_)                                              \
  do {                                                                        \
    if (K <= 32) PW_KERNEL(BN_, WR_, WC_, 1, 32);                             \
    else if (K <= 64) PW_KERNEL(BN_, WR_, WC_, 1, 64);                        \
    else PW_KERNEL(BN_, WR_, WC_, 2, 64);                                     \
  } while (0)

  if (K <= 32 && (K & 7) == 0 && N > 64 && N <= 192) {
    // one-k-step expansion: direct fragment loads, no LDS/barriers
    const long long mtiles = (M + 255) / 256;
    const unsigned grid = (unsigned)(mtiles < 8192 ? mtiles : 8192);
    if (N <= 144) {
      if (stats_p)
        pw_gemm_smallk_kernel<9, true><<<grid, 256, 0, stream>>>(xp, wp, yp, stats_p, M, N, K);
      else
        pw_gemm_smallk_kernel<9, false><<<grid, 256, 0, stream>>>(xp, wp, yp, nullptr, M, N, K);
    } else {
      if (stats_p)
        pw_gemm_smallk_kernel<12, true><<<grid, 256, 0, stream>>>(xp, wp, yp, stats_p, M, N, K);
      else
        pw_gemm_smallk_kernel<12, false><<<grid, 256, 0, stream>>>(xp, wp, yp, nullptr, M, N, K);
    }
  } else if (N <= 32) {
    PW_LAUNCH(32, 4, 1);
  } else if (N <= 64) {
    PW_LAUNCH(64, 2, 2);
  } else {
    PW_LAUNCH(128, 2, 2);
  }
#undef PW_LAUNCH
#undef PW_KERNEL
  return y;
}

// dy: (B, C_out, H, W) channels_last; x: (B, C_in, H, W) channels_last.
// Returns dW (C_out, C_in, 1, 1) bf16.
at::Tensor pw_conv2d_bwd_weight_mfma(at::Tensor dy, at::Tensor x) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda(), "pw_wgrad: CUDA tensors required");
  TORCH_CHECK(dy.scalar_type() == at::kBFloat16 && x.scalar_type() == at::kBFloat16,
              "pw_wgrad: bf16 only");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                  x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "pw_wgrad: channels_last required");
  const long long M = (long long)x.size(0) * x.size(2) * x.size(3);
  const int K = (int)x.size(1);
  const int N = (int)dy.size(1);
  TORCH_CHECK(dy.size(0) == x.size(0) && dy.size(2) == x.size(2) &&
                  dy.size(3) == x.size(3),
              "pw_wgrad: shape mismatch");

  const int ntiles = (N + WT - 1) / WT;
  const int ktiles = (K + WT - 1) / WT;
  const long long kn = (long long)ntiles * ktiles;
  // enough m-chunks to fill the chip, but >=8 stage iterations per block
  long long chunks = kMaxGrid / kn;
  const long long max_chunks = (M + 8 * TM - 1) / (8 * TM);
  if (chunks > max_chunks) chunks = max_chunks;
  if (chunks < 1) chunks = 1;
  if (chunks > 2048) chunks = 2048;
  const long long rows_per_chunk =
      ((M + chunks - 1) / chunks + TM - 1) / TM * TM;  // TM-aligned
  chunks = (M + rows_per_chunk - 1) / rows_per_chunk;

  auto stream = at::hip::getCurrentHIPStream().stream();
  auto part = at::empty({chunks, (long long)N, (long long)K},
                        x.options().dtype(at::kFloat));
  dim3 grid((unsigned)kn, (unsigned)chunks);
  pw_wgrad_kernel<<<grid, 256, 0, stream>>>(
      (const __hip_bfloat16*)dy.data_ptr(), (const __hip_bfloat16*)x.data_ptr(),
      part.data_ptr<float>(), M, N, K, rows_per_chunk);

  auto dw = at::empty({(long long)N, (long long)K, 1, 1}, dy.options());
  const long long nk = (long long)N * K;
  pw_wgrad_reduce_kernel<<<dim3((unsigned)((nk + 255) / 256)), 256, 0, stream>>>(
      part.data_ptr<float>(), (__hip_bfloat16*)dw.data_ptr(), nk, (int)chunks);
  return dw;
}
