// Stem convolution (kxk stride-2, few input channels) as an implicit-GEMM
// on MFMA matrix cores, NHWC, gfx950.
//
//   y[M, N] = patch[M, K] @ w[N, K]^T
//     M = B*Ho*Wo, N = C_out, K = KH*KW*C_in (27 for RGB k3, 108 for the
//     12-chan deepfake_v4 stem), padded to a multiple of 32 (one MFMA
//     k-step) with zeros on the weight side.
//
// The reference delegates the stem to cuDNN (efficientnet.py:275 via
// create_conv2d); on MI355X MIOpen's igemm served it — the last MIOpen
// kernels in the training step (r02 profile) plus ~30 s of find time per
// fresh process. Here:
//   fwd  — B (repacked, zero-padded weights) fragments load directly from
//          global; A fragments GATHER per element (each lane's 8-element
//          fragment spans (kh,kw,c) cells of its output pixel's patch);
//          4 waves stack M (32 rows each), grid (m-tiles, 64-col n-tiles).
//          Optional BN-stats epilogue (bucketed, like ops/hip/pwconv.hip).
//   bwd-weight — split-M chunks: dy^T staged transposed through LDS (as in
//          pw_wgrad), patch columns gathered scalar into the [k][m] tile,
//          fp32 partials + reduce. No atomics.
//
// Weight repack (host, tiny): (N, C_in, KH, KW) -> [N][Kpad] with
// k = (kh*KW + kw)*C_in + c, zero-padded to Kpad.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <type_traits>

#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int kStatsBucketsS = 64;

struct StemGeom {
  int B, Cin, H, W, Ho, Wo, KH, KW, N, Kpad;
  int sh, sw, ph, pw;
};

// decode helpers: the k-triples (c, kw, kh) are loop-invariant per thread —
// hoist them out of the m-loop (packed c | kw<<8 | kh<<16, or -1 for the
// zero-pad tail); the (b, ho, wo) decode happens once per output row.
DFD_DEV int stem_ktrip(const StemGeom& g, int k) {
  if (k >= g.KH * g.KW * g.Cin) return -1;
  const int c = k % g.Cin;
  const int cell = k / g.Cin;
  return c | ((cell % g.KW) << 8) | ((cell / g.KW) << 16);
}

struct StemRow {
  int b, hi0, wi0;  // input-space origin of the output pixel's patch
};

DFD_DEV StemRow stem_row(const StemGeom& g, long long m) {
  StemRow r;
  const unsigned mu = (unsigned)m;  // M < 2^32 for any realistic batch
  const unsigned wo = mu % (unsigned)g.Wo;
  const unsigned t = mu / (unsigned)g.Wo;
  const unsigned ho = t % (unsigned)g.Ho;
  r.b = (int)(t / (unsigned)g.Ho);
  r.hi0 = (int)ho * g.sh - g.ph;
  r.wi0 = (int)wo * g.sw - g.pw;
  return r;
}

template <typename T>
DFD_DEV float stem_patch_el(const T* __restrict__ x, const StemGeom& g,
                            const StemRow& r, int trip) {
  if (trip < 0) return 0.f;
  const int hi = r.hi0 + ((trip >> 16) & 0xff);
  const int wi = r.wi0 + ((trip >> 8) & 0xff);
  if (hi < 0 || hi >= g.H || wi < 0 || wi >= g.W) return 0.f;
  return DfdCvt<T>::to_f32(
      x[(((long long)r.b * g.H + hi) * g.W + wi) * g.Cin + (trip & 0xff)]);
}

// ---------------------------------------------------------------------------
// forward: 4 waves x 32 rows (FI=2), 64-col n-tiles (FJ=4), grid-stride m
// ---------------------------------------------------------------------------
template <typename T, bool STATS>
__global__ __launch_bounds__(256) void stem_fwd_kernel(
    const T* __restrict__ x,                // NHWC input
    const __hip_bfloat16* __restrict__ w,   // [N, Kpad] repacked
    T* __restrict__ y,                      // [M, N] (NHWC output)
    float* __restrict__ stats,              // [64, 2, N] or null
    StemGeom g, long long M) {
  constexpr int FI = 2;  // 32 output rows per wave
  constexpr int FJ = 4;  // 64 output cols per block
  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wid = tid / kWave;
  const int lrow = lane & 15;
  const int lk = (lane >> 4) * 8;
  const int n0 = blockIdx.y * 64;

  float ssum[FJ], sq[FJ];
#pragma unroll
  for (int j = 0; j < FJ; ++j) { ssum[j] = 0.f; sq[j] = 0.f; }

  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
  // block M-tile = 4 waves x FI*16 = 128 rows
  const long long mtiles = (M + 127) / 128;
  for (long long mt = blockIdx.x; mt < mtiles; mt += gridDim.x) {
    const long long m0 = mt * 128 + wid * (FI * 16);
    f32x4 acc[FI][FJ];
#pragma unroll
    for (int i = 0; i < FI; ++i)
#pragma unroll
      for (int j = 0; j < FJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    StemRow rows[FI];
#pragma unroll
    for (int i = 0; i < FI; ++i) {
      const long long gm = m0 + i * 16 + lrow;
      rows[i] = stem_row(g, gm < M ? gm : 0);
    }
    for (int k0 = 0; k0 < g.Kpad; k0 += 32) {
      int trips[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) trips[e] = stem_ktrip(g, k0 + lk + e);
      bf16x8 afrag[FI], bfrag[FJ];
#pragma unroll
      for (int i = 0; i < FI; ++i) {
        const long long gm = m0 + i * 16 + lrow;
        afrag[i] = bf16x8{};
        if (gm < M) {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            reinterpret_cast<__bf16*>(&afrag[i])[e] =
                (__bf16)__float2bfloat16(stem_patch_el<T>(x, g, rows[i], trips[e]));
        }
      }
#pragma unroll
      for (int j = 0; j < FJ; ++j) {
        const int gn = n0 + j * 16 + lrow;
        bfrag[j] = bf16x8{};
        if (gn < g.N)
          bfrag[j] = *reinterpret_cast<const bf16x8*>(
              w + (long long)gn * g.Kpad + k0 + lk);
      }
#pragma unroll
      for (int i = 0; i < FI; ++i)
#pragma unroll
        for (int j = 0; j < FJ; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }

#pragma unroll
    for (int i = 0; i < FI; ++i) {
#pragma unroll
      for (int j = 0; j < FJ; ++j) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const long long gm = m0 + i * 16 + crow0 + r;
          const int gn = n0 + j * 16 + ccol;
          if (gm < M && gn < g.N) {
            const T v = DfdCvt<T>::from_f32(acc[i][j][r]);
            y[gm * g.N + gn] = v;
            if (STATS) {
              const float f = DfdCvt<T>::to_f32(v);
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
      float* bucket = stats +
          (size_t)((blockIdx.x * 4 + wid) & (kStatsBucketsS - 1)) * 2 * g.N;
#pragma unroll
      for (int j = 0; j < FJ; ++j) {
        const int gn = n0 + j * 16 + lane;
        if (gn < g.N) {
          atomicAdd(bucket + gn, ssum[j]);
          atomicAdd(bucket + g.N + gn, sq[j]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// bwd-weight: dW[N, K] = dy^T @ patch; 64x64 (n,k) tiles, split-M chunks of
// fp32 partials, patch columns gathered scalar into the [k][m] LDS tile.
// ---------------------------------------------------------------------------
constexpr int SWT = 64;
constexpr int STM = 64;
constexpr int SLDM = STM + 8;

template <typename T>
__global__ __launch_bounds__(256) void stem_wgrad_kernel(
    const T* __restrict__ dy,  // [M, N]
    const T* __restrict__ x,   // NHWC input
    float* __restrict__ part,  // [chunks, N, Kpad]
    StemGeom g, long long M, long long rows_per_chunk) {
  __shared__ __bf16 dyt[SWT * SLDM];  // [n][m]
  __shared__ __bf16 xt[SWT * SLDM];   // [k][m]

  const int ktiles = (g.Kpad + SWT - 1) / SWT;
  const int n0 = (blockIdx.x / ktiles) * SWT;
  const int k0 = (blockIdx.x % ktiles) * SWT;
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

  const int sm = tid & 63;       // m-local
  const int sq_ = tid >> 6;      // 16-col group

  for (long long m0 = r0; m0 < r1; m0 += STM) {
    const long long gm = m0 + sm;
    // dy tile: vec8 along n, transposed into [n][m]
    {
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const int n = n0 + sq_ * 16 + h * 8;
        bf16x8 v = bf16x8{};
        if (std::is_same<T, __hip_bfloat16>::value && gm < r1 &&
            (g.N & 7) == 0 && n + 7 < g.N) {
          v = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const __hip_bfloat16*>(dy) + gm * g.N + n);
        } else if (gm < r1) {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            reinterpret_cast<__bf16*>(&v)[e] =
                (n + e) < g.N
                    ? (__bf16)__float2bfloat16(DfdCvt<T>::to_f32(dy[gm * g.N + n + e]))
                    : (__bf16)0.f;
        }
#pragma unroll
        for (int e = 0; e < 8; ++e)
          dyt[(sq_ * 16 + h * 8 + e) * SLDM + sm] = reinterpret_cast<__bf16*>(&v)[e];
      }
    }
    // patch tile: gathered scalar into [k][m] (k-triples hoisted, row
    // decode once per m)
    {
      const StemRow row = stem_row(g, gm < r1 ? gm : 0);
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const int kbase = k0 + sq_ * 16 + h * 8;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = 0.f;
          if (gm < r1) f = stem_patch_el<T>(x, g, row, stem_ktrip(g, kbase + e));
          xt[(sq_ * 16 + h * 8 + e) * SLDM + sm] = (__bf16)__float2bfloat16(f);
        }
      }
    }
    __syncthreads();

#pragma unroll
    for (int s = 0; s < 2; ++s) {
      bf16x8 afrag[2], bfrag[2];
#pragma unroll
      for (int i = 0; i < 2; ++i)
        afrag[i] = *reinterpret_cast<const bf16x8*>(
            &dyt[(wn + i * 16 + lrow) * SLDM + s * 32 + lk]);
#pragma unroll
      for (int j = 0; j < 2; ++j)
        bfrag[j] = *reinterpret_cast<const bf16x8*>(
            &xt[(wk + j * 16 + lrow) * SLDM + s * 32 + lk]);
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  float* out = part + (size_t)blockIdx.y * g.N * g.Kpad;
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gn = n0 + wn + i * 16 + crow0 + r;
        const int gk = k0 + wk + j * 16 + ccol;
        if (gn < g.N && gk < g.Kpad) out[(size_t)gn * g.Kpad + gk] = acc[i][j][r];
      }
}

__global__ void stem_wgrad_reduce_kernel(const float* __restrict__ part,
                                         float* __restrict__ dw, long long nk,
                                         int chunks) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nk) return;
  float s = 0.f;
  for (int c = 0; c < chunks; ++c) s += part[(size_t)c * nk + i];
  dw[i] = s;
}

StemGeom make_stem_geom(const at::Tensor& x, int N, int KH, int KW, int sh, int sw,
                        int ph, int pw) {
  StemGeom g;
  g.B = (int)x.size(0);
  g.Cin = (int)x.size(1);
  g.H = (int)x.size(2);
  g.W = (int)x.size(3);
  g.KH = KH;
  g.KW = KW;
  g.N = N;
  g.sh = sh;
  g.sw = sw;
  g.ph = ph;
  g.pw = pw;
  g.Ho = (g.H + 2 * ph - KH) / sh + 1;
  g.Wo = (g.W + 2 * pw - KW) / sw + 1;
  g.Kpad = (KH * KW * g.Cin + 31) / 32 * 32;
  return g;
}

}  // namespace

// x: (B, Cin, H, W) channels_last; w_packed: [N, Kpad] bf16 (host-repacked).
// Returns NHWC y (and fills stats [64, 2, N] fp32 zeroed, if provided).
at::Tensor stem_conv2d_fwd(at::Tensor x, at::Tensor w_packed, int64_t n_out,
                           int64_t kh, int64_t kw, int64_t sh, int64_t sw,
                           int64_t ph, int64_t pw,
                           c10::optional<at::Tensor> stats_opt) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "stem: 4D CUDA input required");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "stem: channels_last input required");
  TORCH_CHECK(w_packed.scalar_type() == at::kBFloat16 && w_packed.is_contiguous(),
              "stem: packed bf16 weight required");
  auto g = make_stem_geom(x, (int)n_out, (int)kh, (int)kw, (int)sh, (int)sw,
                          (int)ph, (int)pw);
  TORCH_CHECK(w_packed.numel() == (long long)g.N * g.Kpad, "stem: weight size");
  const long long M = (long long)g.B * g.Ho * g.Wo;
  auto y = at::empty({(long long)g.B, (long long)g.N, (long long)g.Ho, (long long)g.Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto stream = at::hip::getCurrentHIPStream().stream();
  const long long mtiles = (M + 127) / 128;
  dim3 grid((unsigned)(mtiles < 8192 ? mtiles : 8192), (g.N + 63) / 64);
  float* stats_p = nullptr;
  if (stats_opt.has_value()) {
    at::Tensor stats = *stats_opt;
    TORCH_CHECK(stats.scalar_type() == at::kFloat && stats.is_contiguous() &&
                    stats.numel() == (long long)kStatsBucketsS * 2 * g.N,
                "stem: stats must be fp32 [64, 2, N]");
    stats_p = stats.data_ptr<float>();
  }
#define STEM_FWD_T(T)                                                         \
  do {                                                                        \
    if (stats_p)                                                              \
      stem_fwd_kernel<T, true><<<grid, 256, 0, stream>>>(                     \
          (const T*)x.data_ptr(), (const __hip_bfloat16*)w_packed.data_ptr(), \
          (T*)y.data_ptr(), stats_p, g, M);                                   \
    else                                                                      \
      stem_fwd_kernel<T, false><<<grid, 256, 0, stream>>>(                    \
          (const T*)x.data_ptr(), (const __hip_bfloat16*)w_packed.data_ptr(), \
          (T*)y.data_ptr(), nullptr, g, M);                                   \
  } while (0)
  switch (x.scalar_type()) {
    case at::kBFloat16: STEM_FWD_T(__hip_bfloat16); break;
    case at::kHalf: STEM_FWD_T(__half); break;
    case at::kFloat: STEM_FWD_T(float); break;
    default: TORCH_CHECK(false, "stem: unsupported dtype");
  }
#undef STEM_FWD_T
  return y;
}

// dy: (B, N, Ho, Wo) channels_last; x: input. Returns fp32 [N, Kpad] (the
// python wrapper unpacks to (N, Cin, KH, KW)).
at::Tensor stem_conv2d_bwd_weight(at::Tensor dy, at::Tensor x, int64_t kh,
                                  int64_t kw, int64_t sh, int64_t sw, int64_t ph,
                                  int64_t pw) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda(), "stem_wgrad: CUDA tensors required");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                  x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "stem_wgrad: channels_last required");
  auto g = make_stem_geom(x, (int)dy.size(1), (int)kh, (int)kw, (int)sh, (int)sw,
                          (int)ph, (int)pw);
  TORCH_CHECK(dy.size(2) == g.Ho && dy.size(3) == g.Wo, "stem_wgrad: dy shape");
  const long long M = (long long)g.B * g.Ho * g.Wo;

  const int ntiles = (g.N + SWT - 1) / SWT;
  const int ktiles = (g.Kpad + SWT - 1) / SWT;
  const long long kn = (long long)ntiles * ktiles;
  long long chunks = kMaxGrid / kn;
  const long long max_chunks = (M + 8 * STM - 1) / (8 * STM);
  if (chunks > max_chunks) chunks = max_chunks;
  if (chunks < 1) chunks = 1;
  if (chunks > 2048) chunks = 2048;
  const long long rows_per_chunk =
      ((M + chunks - 1) / chunks + STM - 1) / STM * STM;
  chunks = (M + rows_per_chunk - 1) / rows_per_chunk;

  auto stream = at::hip::getCurrentHIPStream().stream();
  auto part = at::empty({chunks, (long long)g.N, (long long)g.Kpad},
                        x.options().dtype(at::kFloat));
  dim3 grid((unsigned)kn, (unsigned)chunks);
#define STEM_WG_T(T)                                                          \
  stem_wgrad_kernel<T><<<grid, 256, 0, stream>>>(                             \
      (const T*)dy.data_ptr(), (const T*)x.data_ptr(), part.data_ptr<float>(),\
      g, M, rows_per_chunk)
  switch (x.scalar_type()) {
    case at::kBFloat16: STEM_WG_T(__hip_bfloat16); break;
    case at::kHalf: STEM_WG_T(__half); break;
    case at::kFloat: STEM_WG_T(float); break;
    default: TORCH_CHECK(false, "stem_wgrad: unsupported dtype");
  }
#undef STEM_WG_T
  auto dw = at::empty({(long long)g.N, (long long)g.Kpad},
                      x.options().dtype(at::kFloat));
  const long long nk = (long long)g.N * g.Kpad;
  stem_wgrad_reduce_kernel<<<dim3((unsigned)((nk + 255) / 256)), 256, 0, stream>>>(
      part.data_ptr<float>(), dw.data_ptr<float>(), nk, (int)chunks);
  return dw;
}
