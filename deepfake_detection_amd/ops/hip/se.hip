// Fused squeeze-excite chain, NHWC, gfx950.
//
// Forward = ONE kernel per batch doing pool -> 1x1 reduce (GEMV) -> SiLU ->
// 1x1 expand (GEMV) -> sigmoid gate, + one elementwise gate-apply kernel —
// replacing the reference's 6-kernel chain per SE block
// (reference efficientnet_blocks.py:104-110). Intermediates (s, z1, r, g)
// are saved for backward; the tiny dense backward algebra runs in torch
// (ops/se.py), the HW-scale reductions here.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int kMaxCreg = 16;  // C/blockDim ceiling for pooling registers

// grid = N blocks; per block: pool (fp32), reduce GEMV per wave, expand GEMV
// per thread. act on the reduce output is SiLU or ReLU.
template <typename T, Act ACT>
__global__ void se_gate_fwd_kernel(
    const T* __restrict__ x, const float* __restrict__ w1,  // [Cr, C]
    const float* __restrict__ b1, const float* __restrict__ w2,  // [C, Cr]
    const float* __restrict__ b2,
    float* __restrict__ s_out,   // [N, C] pooled
    float* __restrict__ z1_out,  // [N, Cr] pre-act
    float* __restrict__ r_out,   // [N, Cr] post-act
    float* __restrict__ g_out,   // [N, C] sigmoid gate
    int C, int Cr, long long HW) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* s_lds = reinterpret_cast<float*>(smem_raw);        // [C]
  float* r_lds = s_lds + C;                                 // [Cr]

  const long long n = blockIdx.x;
  const T* xn = x + n * HW * C;
  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wid = tid / kWave;
  const int nw = blockDim.x / kWave;

  // ---- pool: thread t accumulates channels t, t+B, ... over all rows
  float acc[kMaxCreg];
  const int nreg = (C + blockDim.x - 1) / blockDim.x;
#pragma unroll 4
  for (int j = 0; j < nreg; ++j) acc[j] = 0.f;
  for (long long s = 0; s < HW; ++s) {
    const T* row = xn + s * C;
    for (int j = 0; j < nreg; ++j) {
      const int c = tid + j * blockDim.x;
      if (c < C) acc[j] += DfdCvt<T>::to_f32(row[c]);
    }
  }
  const float invHW = 1.f / (float)HW;
  for (int j = 0; j < nreg; ++j) {
    const int c = tid + j * blockDim.x;
    if (c < C) {
      const float m = acc[j] * invHW;
      s_lds[c] = m;
      s_out[n * C + c] = m;
    }
  }
  __syncthreads();

  // ---- reduce GEMV: z1[j] = w1[j,:] . s + b1[j]; one wave per j
  for (int j = wid; j < Cr; j += nw) {
    const float* wrow = w1 + (long long)j * C;
    float d = 0.f;
    for (int c = lane; c < C; c += kWave) d += wrow[c] * s_lds[c];
    d = wave_sum(d);
    if (lane == 0) {
      const float z1 = d + (b1 ? b1[j] : 0.f);
      const float r = act_fwd(z1, ACT);
      z1_out[n * Cr + j] = z1;
      r_out[n * Cr + j] = r;
      r_lds[j] = r;
    }
  }
  __syncthreads();

  // ---- expand GEMV: g[c] = sigmoid(w2[c,:] . r + b2[c]); one thread per c
  for (int c = tid; c < C; c += blockDim.x) {
    const float* wrow = w2 + (long long)c * Cr;
    float d = b2 ? b2[c] : 0.f;
    for (int j = 0; j < Cr; ++j) d += wrow[j] * r_lds[j];
    g_out[n * C + c] = 1.f / (1.f + __expf(-d));
  }
}

// y = x * g (broadcast over HW)
template <typename T>
__global__ void gate_apply_kernel(const T* __restrict__ x, const float* __restrict__ g,
                                  T* __restrict__ y, long long total, int C, long long HW) {
  const long long idx0 = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (long long i = idx0; i < total; i += stride) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long long k = i + j;
      if (k >= total) break;
      const int c = (int)(k % C);
      const long long n = k / ((long long)C * HW);
      y[k] = DfdCvt<T>::from_f32(DfdCvt<T>::to_f32(x[k]) * g[n * C + c]);
    }
  }
}

// backward pass 1: dx = dy * g (elementwise)  and  dg[n,c] = sum_hw dy*x
template <typename T, int VEC>
__global__ void se_bwd_reduce_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const float* __restrict__ g,
    T* __restrict__ dx, float* __restrict__ dg, int C, long long HW) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int nw = blockDim.x / kWave;
  const long long n = blockIdx.x;
  const int c0 = blockIdx.y * kWave * VEC + lane * VEC;
  if (c0 >= C) return;

  const T* xn = x + n * HW * C;
  const T* dn = dy + n * HW * C;
  T* dxn = dx + n * HW * C;

  float acc[VEC], gv[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    acc[j] = 0.f;
    gv[j] = (c0 + j < C) ? g[n * C + c0 + j] : 0.f;
  }

  for (long long s = wid; s < HW; s += nw) {
    const long long base = s * C + c0;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      if (c0 + j < C) {
        const float xv = DfdCvt<T>::to_f32(xn[base + j]);
        const float dv = DfdCvt<T>::to_f32(dn[base + j]);
        acc[j] += dv * xv;
        dxn[base + j] = DfdCvt<T>::from_f32(dv * gv[j]);
      }
    }
  }

  __shared__ float lds[4][kWave];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    __syncthreads();
    lds[wid][lane] = acc[j];
    __syncthreads();
    if (wid == 0) {
      float t = 0.f;
      for (int w = 0; w < nw; ++w) t += lds[w][lane];
      if (c0 + j < C) dg[n * C + c0 + j] = t;
    }
  }
}

// backward pass 2: dx += ds[n,c] / HW
template <typename T>
__global__ void se_bwd_add_pool_kernel(T* __restrict__ dx, const float* __restrict__ ds,
                                       long long total, int C, long long HW) {
  const long long idx0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const float inv = 1.f / (float)HW;
  for (long long k = idx0; k < total; k += stride) {
    const int c = (int)(k % C);
    const long long n = k / ((long long)C * HW);
    dx[k] = DfdCvt<T>::from_f32(DfdCvt<T>::to_f32(dx[k]) + ds[n * C + c] * inv);
  }
}

}  // namespace

// Returns {y, s, z1, r, g}
std::vector<at::Tensor> se_fwd(at::Tensor x, at::Tensor w1, at::Tensor b1,
                               at::Tensor w2, at::Tensor b2, std::string act_s) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "se_fwd: 4D CUDA tensor expected");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "se_fwd: channels_last input required");
  const long long N = x.size(0);
  const int C = (int)x.size(1);
  const long long HW = (long long)x.size(2) * x.size(3);
  const int Cr = (int)w1.size(0);
  TORCH_CHECK((int)w1.size(1) == C && (int)w2.size(0) == C && (int)w2.size(1) == Cr,
              "se_fwd: weight shape mismatch");
  TORCH_CHECK(C <= 256 * kMaxCreg, "se_fwd: C too large");

  auto w1c = w1.to(at::kFloat).contiguous();
  auto w2c = w2.to(at::kFloat).contiguous();
  auto b1c = b1.defined() ? b1.to(at::kFloat).contiguous() : at::Tensor();
  auto b2c = b2.defined() ? b2.to(at::kFloat).contiguous() : at::Tensor();

  auto opts_f = x.options().dtype(at::kFloat);
  auto s = at::empty({N, (long long)C}, opts_f);
  auto z1 = at::empty({N, (long long)Cr}, opts_f);
  auto r = at::empty({N, (long long)Cr}, opts_f);
  auto g = at::empty({N, (long long)C}, opts_f);
  auto y = at::empty_like(x);

  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const int lds_bytes = (C + Cr) * sizeof(float);
  const bool relu = act_s == "relu";

#define LAUNCH_FWD(T)                                                          \
  do {                                                                         \
    if (relu)                                                                  \
      hipLaunchKernelGGL((se_gate_fwd_kernel<T, Act::kRelu>), dim3((unsigned)N), \
                         dim3(block), lds_bytes, stream, (const T*)x.data_ptr(), \
                         w1c.data_ptr<float>(),                                \
                         b1c.defined() ? b1c.data_ptr<float>() : nullptr,      \
                         w2c.data_ptr<float>(),                                \
                         b2c.defined() ? b2c.data_ptr<float>() : nullptr,      \
                         s.data_ptr<float>(), z1.data_ptr<float>(),            \
                         r.data_ptr<float>(), g.data_ptr<float>(), C, Cr, HW); \
    else                                                                       \
      hipLaunchKernelGGL((se_gate_fwd_kernel<T, Act::kSilu>), dim3((unsigned)N), \
                         dim3(block), lds_bytes, stream, (const T*)x.data_ptr(), \
                         w1c.data_ptr<float>(),                                \
                         b1c.defined() ? b1c.data_ptr<float>() : nullptr,      \
                         w2c.data_ptr<float>(),                                \
                         b2c.defined() ? b2c.data_ptr<float>() : nullptr,      \
                         s.data_ptr<float>(), z1.data_ptr<float>(),            \
                         r.data_ptr<float>(), g.data_ptr<float>(), C, Cr, HW); \
    const long long total = N * C * HW;                                        \
    hipLaunchKernelGGL((gate_apply_kernel<T>), dim3(dfd_grid(total / 4 + 1, block)), \
                       dim3(block), 0, stream, (const T*)x.data_ptr(),         \
                       g.data_ptr<float>(), (T*)y.data_ptr(), total, C, HW);   \
  } while (0)

  if (x.scalar_type() == at::kBFloat16) LAUNCH_FWD(__hip_bfloat16);
  else if (x.scalar_type() == at::kHalf) LAUNCH_FWD(__half);
  else if (x.scalar_type() == at::kFloat) LAUNCH_FWD(float);
  else TORCH_CHECK(false, "se_fwd: unsupported dtype");
#undef LAUNCH_FWD

  return {y, s, z1, r, g};
}

// pass 1 of backward: returns {dx (gate-direct part), dg}
std::vector<at::Tensor> se_bwd_reduce(at::Tensor dy, at::Tensor x, at::Tensor g) {
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  const long long N = x.size(0);
  const int C = (int)x.size(1);
  const long long HW = (long long)x.size(2) * x.size(3);
  auto dx = at::empty_like(x);
  auto dg = at::empty({N, (long long)C}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  constexpr int VEC = 4;
  dim3 grid((unsigned)N, (C + kWave * VEC - 1) / (kWave * VEC));

#define LAUNCH(T)                                                              \
  hipLaunchKernelGGL((se_bwd_reduce_kernel<T, VEC>), grid, dim3(256), 0, stream, \
                     (const T*)dy.data_ptr(), (const T*)x.data_ptr(),          \
                     g.data_ptr<float>(), (T*)dx.data_ptr(), dg.data_ptr<float>(), C, HW)
  if (x.scalar_type() == at::kBFloat16) LAUNCH(__hip_bfloat16);
  else if (x.scalar_type() == at::kHalf) LAUNCH(__half);
  else if (x.scalar_type() == at::kFloat) LAUNCH(float);
  else TORCH_CHECK(false, "se_bwd_reduce: unsupported dtype");
#undef LAUNCH
  return {dx, dg};
}

// pass 2: dx += ds/HW in place
void se_bwd_add_pool(at::Tensor dx, at::Tensor ds) {
  const long long N = dx.size(0);
  const int C = (int)dx.size(1);
  const long long HW = (long long)dx.size(2) * dx.size(3);
  const long long total = N * C * HW;
  auto dsc = ds.contiguous();
  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const int grid = dfd_grid(total, block);
#define LAUNCH(T)                                                              \
  hipLaunchKernelGGL((se_bwd_add_pool_kernel<T>), dim3(grid), dim3(block), 0,  \
                     stream, (T*)dx.data_ptr(), dsc.data_ptr<float>(), total, C, HW)
  if (dx.scalar_type() == at::kBFloat16) LAUNCH(__hip_bfloat16);
  else if (dx.scalar_type() == at::kHalf) LAUNCH(__half);
  else if (dx.scalar_type() == at::kFloat) LAUNCH(float);
  else TORCH_CHECK(false, "se_bwd_add_pool: unsupported dtype");
#undef LAUNCH
}
