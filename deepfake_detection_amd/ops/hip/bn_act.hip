// Fused BatchNorm2d + activation (forward + backward), NHWC, gfx950.
//
// Replaces the reference's separate cuDNN BatchNorm + jit-scripted Swish
// kernels (reference dfd/timm/models/layers/activations.py:19-48; BN at
// every efficientnet block). Input/activation dtype
// bf16/fp16/fp32; all statistics and parameters fp32.
//
// Layout: channels_last (N,C,H,W) == row-major [M, C] with M = N*H*W and C
// contiguous. Per-channel reductions: each wave owns a 64*VEC-channel slab,
// lanes read ushort4 (8 B) vectors; 4 waves per block cover different rows;
// per-block partials combine in LDS and one atomicAdd per channel publishes
// to the fp32 accumulator (Guideline 12).
//
// Training forward is two passes (stats reduce -> finalize -> fused
// normalize+act elementwise); backward is a reduce pass (dgamma/dbeta with
// act' recompute) + an elementwise dx pass. SiLU backward recomputes
// sigma(z) from the saved mean/invstd (the "hard part" flagged in
// SURVEY.md §7).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// ---------------------------------------------------------------------------
// stats reduce: sum and sumsq per channel
// ---------------------------------------------------------------------------
template <typename T, int VEC>
__global__ void bn_stats_kernel(const T* __restrict__ x, float* __restrict__ sum,
                                float* __restrict__ sumsq, long long M, int C) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int nw = blockDim.x / kWave;

  const int cb = blockIdx.x;              // channel slab: 64*VEC channels
  const int c0 = cb * kWave * VEC + lane * VEC;
  if (c0 >= C) return;

  float s[VEC], q[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) { s[j] = 0.f; q[j] = 0.f; }

  const bool full = (c0 + VEC) <= C;
  const long long row0 = (long long)blockIdx.y * nw + wid;
  const long long rstride = (long long)gridDim.y * nw;

  for (long long r = row0; r < M; r += rstride) {
    const T* px = x + r * C + c0;
    if (full && VEC == 4 && (sizeof(T) == 2)) {
      ushort4 v = *reinterpret_cast<const ushort4*>(px);
      const T* e = reinterpret_cast<const T*>(&v);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float f = DfdCvt<T>::to_f32(e[j]);
        s[j] += f; q[j] += f * f;
      }
    } else {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        if (c0 + j < C) {
          float f = DfdCvt<T>::to_f32(px[j]);
          s[j] += f; q[j] += f * f;
        }
      }
    }
  }

  // combine the block's waves per channel through LDS
  __shared__ float lds[4][kWave];  // one VEC element at a time
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    __syncthreads();
    lds[wid][lane] = s[j];
    __syncthreads();
    if (wid == 0) {
      float acc = 0.f;
      for (int w = 0; w < nw; ++w) acc += lds[w][lane];
      if (c0 + j < C && acc != 0.f) atomicAdd(&sum[c0 + j], acc);
      else if (c0 + j < C) atomicAdd(&sum[c0 + j], acc);
    }
    __syncthreads();
    lds[wid][lane] = q[j];
    __syncthreads();
    if (wid == 0) {
      float acc = 0.f;
      for (int w = 0; w < nw; ++w) acc += lds[w][lane];
      if (c0 + j < C) atomicAdd(&sumsq[c0 + j], acc);
    }
  }
}

// ---------------------------------------------------------------------------
// finalize: stats -> mean/invstd (+ running update) -> scale/shift
// ---------------------------------------------------------------------------
__global__ void bn_finalize_train_kernel(
    const float* __restrict__ sum, const float* __restrict__ sumsq,
    const float* __restrict__ weight, const float* __restrict__ bias,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    float* __restrict__ save_mean, float* __restrict__ save_invstd,
    float* __restrict__ scale, float* __restrict__ shift,
    long long M, int C, float momentum, float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float mean = sum[c] / (float)M;
  float var = sumsq[c] / (float)M - mean * mean;
  if (var < 0.f) var = 0.f;  // numerical guard
  const float invstd = rsqrtf(var + eps);
  save_mean[c] = mean;
  save_invstd[c] = invstd;
  if (running_mean != nullptr) {
    const float unbiased = (M > 1) ? var * (float)M / (float)(M - 1) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
  const float g = weight ? weight[c] : 1.f;
  const float b = bias ? bias[c] : 0.f;
  scale[c] = g * invstd;
  shift[c] = b - mean * g * invstd;
}

__global__ void bn_finalize_eval_kernel(
    const float* __restrict__ running_mean, const float* __restrict__ running_var,
    const float* __restrict__ weight, const float* __restrict__ bias,
    float* __restrict__ save_mean, float* __restrict__ save_invstd,
    float* __restrict__ scale, float* __restrict__ shift, int C, float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float mean = running_mean[c];
  const float invstd = rsqrtf(running_var[c] + eps);
  save_mean[c] = mean;
  save_invstd[c] = invstd;
  const float g = weight ? weight[c] : 1.f;
  const float b = bias ? bias[c] : 0.f;
  scale[c] = g * invstd;
  shift[c] = b - mean * g * invstd;
}

// ---------------------------------------------------------------------------
// fused normalize + act elementwise: y = act(scale*x + shift)
// ---------------------------------------------------------------------------
template <typename T, Act ACT>
__global__ void bn_act_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                  const float* __restrict__ scale,
                                  const float* __restrict__ shift,
                                  long long total, int C) {
  // vectorized: 4 elements (8 B for 16-bit dtypes) per thread
  const long long idx0 = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (long long i = idx0; i < total; i += stride) {
    if (i + 4 <= total && (C % 4 == 0)) {
      const int c = (int)(i % C);
      if (sizeof(T) == 2) {
        ushort4 v = *reinterpret_cast<const ushort4*>(x + i);
        T* e = reinterpret_cast<T*>(&v);
        ushort4 o;
        T* oe = reinterpret_cast<T*>(&o);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float z = fmaf(DfdCvt<T>::to_f32(e[j]), scale[c + j], shift[c + j]);
          oe[j] = DfdCvt<T>::from_f32(act_fwd(z, ACT));
        }
        *reinterpret_cast<ushort4*>(y + i) = o;
      } else {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float z = fmaf(DfdCvt<T>::to_f32(x[i + j]), scale[c + j], shift[c + j]);
          y[i + j] = DfdCvt<T>::from_f32(act_fwd(z, ACT));
        }
      }
    } else {
      for (int j = 0; j < 4 && i + j < total; ++j) {
        const int c = (int)((i + j) % C);
        const float z = fmaf(DfdCvt<T>::to_f32(x[i + j]), scale[c], shift[c]);
        y[i + j] = DfdCvt<T>::from_f32(act_fwd(z, ACT));
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward reduce: dbeta = sum g, dgamma = sum g*xhat, g = dy*act'(z)
// ---------------------------------------------------------------------------
template <typename T, Act ACT, int VEC>
__global__ void bn_act_bwd_reduce_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ bias,
    float* __restrict__ dgamma, float* __restrict__ dbeta,
    long long M, int C) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int nw = blockDim.x / kWave;

  const int c0 = blockIdx.x * kWave * VEC + lane * VEC;
  if (c0 >= C) return;

  float sg[VEC], sgx[VEC];
  float mn[VEC], is[VEC], ga[VEC], be[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    sg[j] = 0.f; sgx[j] = 0.f;
    const int c = min(c0 + j, C - 1);
    mn[j] = mean[c]; is[j] = invstd[c];
    ga[j] = weight ? weight[c] : 1.f;
    be[j] = bias ? bias[c] : 0.f;
  }

  const bool full = (c0 + VEC) <= C;
  const long long row0 = (long long)blockIdx.y * nw + wid;
  const long long rstride = (long long)gridDim.y * nw;

  for (long long r = row0; r < M; r += rstride) {
    const T* px = x + r * C + c0;
    const T* pd = dy + r * C + c0;
    if (full && VEC == 4 && sizeof(T) == 2) {
      ushort4 vx = *reinterpret_cast<const ushort4*>(px);
      ushort4 vd = *reinterpret_cast<const ushort4*>(pd);
      const T* ex = reinterpret_cast<const T*>(&vx);
      const T* ed = reinterpret_cast<const T*>(&vd);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float xf = DfdCvt<T>::to_f32(ex[j]);
        const float xh = (xf - mn[j]) * is[j];
        const float z = fmaf(ga[j], xh, be[j]);
        const float g = DfdCvt<T>::to_f32(ed[j]) * act_bwd(z, ACT);
        sg[j] += g; sgx[j] += g * xh;
      }
    } else {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        if (c0 + j < C) {
          const float xf = DfdCvt<T>::to_f32(px[j]);
          const float xh = (xf - mn[j]) * is[j];
          const float z = fmaf(ga[j], xh, be[j]);
          const float g = DfdCvt<T>::to_f32(pd[j]) * act_bwd(z, ACT);
          sg[j] += g; sgx[j] += g * xh;
        }
      }
    }
  }

  __shared__ float lds[4][kWave];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    __syncthreads();
    lds[wid][lane] = sg[j];
    __syncthreads();
    if (wid == 0) {
      float acc = 0.f;
      for (int w = 0; w < nw; ++w) acc += lds[w][lane];
      if (c0 + j < C) atomicAdd(&dbeta[c0 + j], acc);
    }
    __syncthreads();
    lds[wid][lane] = sgx[j];
    __syncthreads();
    if (wid == 0) {
      float acc = 0.f;
      for (int w = 0; w < nw; ++w) acc += lds[w][lane];
      if (c0 + j < C) atomicAdd(&dgamma[c0 + j], acc);
    }
  }
}

// ---------------------------------------------------------------------------
// backward dx elementwise:
//   train: dx = gamma*invstd * (g - dbeta/M - xhat*dgamma/M)
//   eval:  dx = gamma*invstd * g
// ---------------------------------------------------------------------------
template <typename T, int N>
struct alignas(sizeof(T) * N) BVec {
  T v[N];
};

template <typename T, Act ACT, bool TRAIN, int VEC>
__global__ void bn_act_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ bias,
    const float* __restrict__ dgamma, const float* __restrict__ dbeta,
    long long M, int C, float invM) {
  const int cv = C / VEC;
  const long long total = M * cv;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(idx % cv) * VEC;
    const long long r = idx / cv;
    const BVec<T, VEC> xv = *reinterpret_cast<const BVec<T, VEC>*>(x + r * C + c);
    const BVec<T, VEC> dv = *reinterpret_cast<const BVec<T, VEC>*>(dy + r * C + c);
    BVec<T, VEC> ov;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      const float is = invstd[c + j];
      const float xh = (DfdCvt<T>::to_f32(xv.v[j]) - mean[c + j]) * is;
      const float ga = weight ? weight[c + j] : 1.f;
      const float z = fmaf(ga, xh, bias ? bias[c + j] : 0.f);
      const float g = DfdCvt<T>::to_f32(dv.v[j]) * act_bwd(z, ACT);
      float v;
      if (TRAIN) {
        v = ga * is * (g - dbeta[c + j] * invM - xh * dgamma[c + j] * invM);
      } else {
        v = ga * is * g;
      }
      ov.v[j] = DfdCvt<T>::from_f32(v);
    }
    *reinterpret_cast<BVec<T, VEC>*>(dx + r * C + c) = ov;
  }
}

Act act_from_string(const std::string& s) {
  if (s == "silu") return Act::kSilu;
  if (s == "relu") return Act::kRelu;
  TORCH_CHECK(s == "none", "unknown act: ", s);
  return Act::kNone;
}

#define DISPATCH_DTYPE(scalar_type, NAME, ...)                        \
  [&] {                                                               \
    if (scalar_type == at::kBFloat16) {                               \
      using T = __hip_bfloat16;                                       \
      return __VA_ARGS__();                                           \
    } else if (scalar_type == at::kHalf) {                            \
      using T = __half;                                               \
      return __VA_ARGS__();                                           \
    } else if (scalar_type == at::kFloat) {                           \
      using T = float;                                                \
      return __VA_ARGS__();                                           \
    } else {                                                          \
      TORCH_CHECK(false, NAME ": unsupported dtype");                 \
    }                                                                 \
  }()

#define DISPATCH_ACT(act, ...)                                        \
  [&] {                                                               \
    if (act == Act::kSilu) {                                          \
      constexpr Act ACT = Act::kSilu;                                 \
      return __VA_ARGS__();                                           \
    } else if (act == Act::kRelu) {                                   \
      constexpr Act ACT = Act::kRelu;                                 \
      return __VA_ARGS__();                                           \
    } else {                                                          \
      constexpr Act ACT = Act::kNone;                                 \
      return __VA_ARGS__();                                           \
    }                                                                 \
  }()

}  // namespace

// x: (N,C,H,W) channels_last. Returns {y, save_mean, save_invstd}.
std::vector<at::Tensor> bn_act_fwd(
    at::Tensor x, at::Tensor weight, at::Tensor bias,
    at::Tensor running_mean, at::Tensor running_var,
    bool training, double momentum, double eps, std::string act_s) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "bn_act_fwd: 4D CUDA tensor expected");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn_act_fwd: channels_last input required");
  const Act act = act_from_string(act_s);
  const int C = (int)x.size(1);
  const long long M = (long long)x.size(0) * x.size(2) * x.size(3);
  const long long total = M * C;

  auto stream = at::cuda::getCurrentHIPStream();
  auto opts_f = x.options().dtype(at::kFloat);
  auto y = at::empty_like(x);
  auto save_mean = at::empty({C}, opts_f);
  auto save_invstd = at::empty({C}, opts_f);
  auto scale = at::empty({C}, opts_f);
  auto shift = at::empty({C}, opts_f);

  const float* w_p = weight.defined() ? weight.data_ptr<float>() : nullptr;
  const float* b_p = bias.defined() ? bias.data_ptr<float>() : nullptr;

  if (training) {
    auto sum = at::zeros({C}, opts_f);
    auto sumsq = at::zeros({C}, opts_f);
    constexpr int VEC = 4;
    const int slabs = (C + kWave * VEC - 1) / (kWave * VEC);
    int gy = dfd_grid(M / 4 + 1, 64, kMaxGrid / slabs);
    dim3 grid(slabs, gy);
    DISPATCH_DTYPE(x.scalar_type(), "bn_stats", [&] {
      hipLaunchKernelGGL((bn_stats_kernel<T, VEC>), grid, dim3(256), 0, stream,
                         (const T*)x.data_ptr(), sum.data_ptr<float>(),
                         sumsq.data_ptr<float>(), M, C);
    });
    hipLaunchKernelGGL(bn_finalize_train_kernel, dim3((C + 255) / 256), dim3(256), 0, stream,
                       sum.data_ptr<float>(), sumsq.data_ptr<float>(), w_p, b_p,
                       running_mean.defined() ? running_mean.data_ptr<float>() : nullptr,
                       running_var.defined() ? running_var.data_ptr<float>() : nullptr,
                       save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       M, C, (float)momentum, (float)eps);
  } else {
    hipLaunchKernelGGL(bn_finalize_eval_kernel, dim3((C + 255) / 256), dim3(256), 0, stream,
                       running_mean.data_ptr<float>(), running_var.data_ptr<float>(),
                       w_p, b_p, save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
                       scale.data_ptr<float>(), shift.data_ptr<float>(), C, (float)eps);
  }

  const int block = 256;
  const int grid_e = dfd_grid(total / 4 + 1, block);
  DISPATCH_DTYPE(x.scalar_type(), "bn_act_fwd", [&] {
    DISPATCH_ACT(act, [&] {
      hipLaunchKernelGGL((bn_act_fwd_kernel<T, ACT>), dim3(grid_e), dim3(block), 0, stream,
                         (const T*)x.data_ptr(), (T*)y.data_ptr(),
                         scale.data_ptr<float>(), shift.data_ptr<float>(), total, C);
    });
  });
  return {y, save_mean, save_invstd};
}

// Returns {dx, dgamma, dbeta}.
std::vector<at::Tensor> bn_act_bwd(
    at::Tensor dy, at::Tensor x, at::Tensor weight, at::Tensor bias,
    at::Tensor save_mean, at::Tensor save_invstd, bool training, std::string act_s) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4, "bn_act_bwd: 4D CUDA tensor expected");
  const Act act = act_from_string(act_s);
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int C = (int)x.size(1);
  const long long M = (long long)x.size(0) * x.size(2) * x.size(3);
  const long long total = M * C;

  auto stream = at::cuda::getCurrentHIPStream();
  auto opts_f = x.options().dtype(at::kFloat);
  auto dgamma = at::zeros({C}, opts_f);
  auto dbeta = at::zeros({C}, opts_f);
  auto dx = at::empty_like(x);

  const float* w_p = weight.defined() ? weight.data_ptr<float>() : nullptr;
  const float* b_p = bias.defined() ? bias.data_ptr<float>() : nullptr;

  constexpr int VEC = 4;
  const int slabs = (C + kWave * VEC - 1) / (kWave * VEC);
  int gy = dfd_grid(M / 4 + 1, 64, kMaxGrid / slabs);
  dim3 grid(slabs, gy);
  DISPATCH_DTYPE(x.scalar_type(), "bn_bwd_reduce", [&] {
    DISPATCH_ACT(act, [&] {
      hipLaunchKernelGGL((bn_act_bwd_reduce_kernel<T, ACT, VEC>), grid, dim3(256), 0, stream,
                         (const T*)dy.data_ptr(), (const T*)x.data_ptr(),
                         save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
                         w_p, b_p, dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), M, C);
    });
  });

  const int block = 256;
  int dxvec = (x.element_size() == 4) ? 4 : 8;
  while (dxvec > 1 && (C % dxvec)) dxvec >>= 1;
  const int grid_e = dfd_grid(M * (C / dxvec), block);
#define DFD_BN_DX(TRAIN, V)                                                       \
  hipLaunchKernelGGL((bn_act_bwd_dx_kernel<T, ACT, TRAIN, V>), dim3(grid_e),      \
                     dim3(block), 0, stream, (const T*)dy.data_ptr(),             \
                     (const T*)x.data_ptr(), (T*)dx.data_ptr(),                   \
                     save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),  \
                     w_p, b_p, dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), \
                     M, C, invM)
  DISPATCH_DTYPE(x.scalar_type(), "bn_bwd_dx", [&] {
    DISPATCH_ACT(act, [&] {
      const float invM = 1.f / (float)M;
      if (training) {
        switch (dxvec) {
          case 8: DFD_BN_DX(true, 8); break;
          case 4: DFD_BN_DX(true, 4); break;
          case 2: DFD_BN_DX(true, 2); break;
          default: DFD_BN_DX(true, 1); break;
        }
      } else {
        switch (dxvec) {
          case 8: DFD_BN_DX(false, 8); break;
          case 4: DFD_BN_DX(false, 4); break;
          case 2: DFD_BN_DX(false, 2); break;
          default: DFD_BN_DX(false, 1); break;
        }
      }
    });
  });
#undef DFD_BN_DX
  return {dx, dgamma, dbeta};
}
