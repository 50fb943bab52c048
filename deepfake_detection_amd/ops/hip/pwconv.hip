// Pointwise (1x1) convolution as an NHWC GEMM on MFMA matrix cores, gfx950.
//
// out[M, N] = x[M, K] @ w[N, K]^T with M = B*H*W, K = C_in, N = C_out —
// the torch conv weight (C_out, C_in, 1, 1) is ALREADY the [N, K]
// B-transposed layout the kernel wants, so there is no repacking.
//
// Tiling (v1, correctness-first):
//   block = 256 threads (4 waves), block tile 128(M) x 128(N)
//   wave  = 64x64 sub-tile = 4x4 fragments of v_mfma_f32_16x16x32_bf16
//   K loop stages A (128x32) and B (128x32, n-major) through LDS with an
//   8-element row pad (bank-conflict-free fragment reads, 16 B each).
//
// Fragment mappings (cdna4 16x16x32 bf16):
//   A/B: row(col) = lane & 15, k = (lane >> 4) * 8 + i   (8 bf16 / lane)
//   C/D: col = lane & 15, row = (lane >> 4) * 4 + reg    (4 fp32 / lane)
//
// Experimental: wired behind DFD_AMD_PW_MFMA=1 (ops/pwconv.py); MIOpen's
// igemm remains the default 1x1 path until this beats it per-shape.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

#if defined(__gfx950__) || defined(__gfx942__) || !defined(__HIP_DEVICE_COMPILE__)

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 128;  // block tile M
constexpr int BN = 128;  // block tile N
constexpr int BK = 32;   // K step (one MFMA K)
constexpr int LDA = BK + 8;   // padded LDS row (elements)

__global__ __launch_bounds__(256) void pw_gemm_bf16_kernel(
    const __hip_bfloat16* __restrict__ x,  // [M, K] row-major
    const __hip_bfloat16* __restrict__ w,  // [N, K] row-major
    __hip_bfloat16* __restrict__ y,        // [M, N] row-major
    long long M, int N, int K) {
  __shared__ __hip_bfloat16 a_lds[BM * LDA];
  __shared__ __hip_bfloat16 b_lds[BN * LDA];

  const long long m0 = (long long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wid = tid / kWave;
  // wave grid 2x2 over the 128x128 block tile
  const int wm = (wid & 1) * 64;   // wave row offset
  const int wn = (wid >> 1) * 64;  // wave col offset

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int lrow = lane & 15;        // fragment row/col
  const int lk = (lane >> 4) * 8;    // fragment k offset

  for (int k0 = 0; k0 < K; k0 += BK) {
    // stage A: 128 rows x 32 k. 256 threads x 2 vec8 loads.
    {
      const int r = tid >> 2;             // 0..63
      const int c = (tid & 3) * 8;        // 0,8,16,24
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int row = r + half * 64;
        const long long gm = m0 + row;
        bf16x8 v = {};
        if (gm < M && k0 + c + 7 < K) {
          v = *reinterpret_cast<const bf16x8*>(x + gm * K + k0 + c);
        } else if (gm < M) {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int kk = k0 + c + e;
            reinterpret_cast<__bf16*>(&v)[e] =
                kk < K ? *reinterpret_cast<const __bf16*>(x + gm * K + kk) : (__bf16)0.f;
          }
        }
        *reinterpret_cast<bf16x8*>(&a_lds[row * LDA + c]) =
            *reinterpret_cast<bf16x8*>(&v);
      }
    }
    // stage B: 128 n-rows x 32 k (w is [N, K] so this is a straight copy)
    {
      const int r = tid >> 2;
      const int c = (tid & 3) * 8;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int row = r + half * 64;
        const int gn = n0 + row;
        bf16x8 v = {};
        if (gn < N && k0 + c + 7 < K) {
          v = *reinterpret_cast<const bf16x8*>(w + (long long)gn * K + k0 + c);
        } else if (gn < N) {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int kk = k0 + c + e;
            reinterpret_cast<__bf16*>(&v)[e] =
                kk < K ? *reinterpret_cast<const __bf16*>(w + (long long)gn * K + kk) : (__bf16)0.f;
          }
        }
        *reinterpret_cast<bf16x8*>(&b_lds[row * LDA + c]) =
            *reinterpret_cast<bf16x8*>(&v);
      }
    }
    __syncthreads();

    // two MFMA K-halves of 8 (BK=32 total... 16x16x32 consumes all 32 at once
    // via the 8-elem fragments at lk and lk+... lk spans (lane>>4)*8 = 0..24)
    bf16x8 afrag[4], bfrag[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      afrag[i] = *reinterpret_cast<const bf16x8*>(&a_lds[(wm + i * 16 + lrow) * LDA + lk]);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bfrag[j] = *reinterpret_cast<const bf16x8*>(&b_lds[(wn + j * 16 + lrow) * LDA + lk]);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    __syncthreads();
  }

  // writeback: C/D col = lane&15, row = (lane>>4)*4 + reg
  const int ccol = lane & 15;
  const int crow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long long gm = m0 + wm + i * 16 + crow0 + r;
        const int gn = n0 + wn + j * 16 + ccol;
        if (gm < M && gn < N)
          y[gm * N + gn] = __float2bfloat16(acc[i][j][r]);
      }
    }
  }
}

}  // namespace

#endif  // gfx950

// x: (B, C_in, H, W) channels_last; w: (C_out, C_in, 1, 1). Returns NHWC y.
at::Tensor pw_conv2d_fwd_mfma(at::Tensor x, at::Tensor w) {
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
#if 1
  auto stream = at::hip::getCurrentHIPStream().stream();
  dim3 grid((unsigned)((M + 127) / 128), (N + 127) / 128);
  pw_gemm_bf16_kernel<<<grid, 256, 0, stream>>>(
      (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)wc.data_ptr(),
      (__hip_bfloat16*)y.data_ptr(), M, N, K);
#endif
  return y;
}
