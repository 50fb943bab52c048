// Prefetcher device op: uint8 NCHW -> {bf16,fp16,fp32} NHWC with fused
// (x - mean)/std normalize. One kernel replaces
// the reference's .half()/.sub_()/.div_() chain + layout change
// (reference dfd/timm/data/loader.py:246-253).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// Each thread produces VEC=4 consecutive NHWC outputs (same spatial s,
// channels c..c+3); input gathers stride HW apart per channel — the uint8
// reads are 1 B each but land in few distinct lines per wave and the L1/L2
// absorb the re-reads across s (inverse-transpose locality).
template <typename T>
__global__ void normalize_u8_nhwc_kernel(
    const unsigned char* __restrict__ x, T* __restrict__ y,
    const float* __restrict__ inv_std, const float* __restrict__ neg_mean_over_std,
    long long total, int C, long long HW) {
  const long long idx0 = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (long long i = idx0; i < total; i += stride) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long long k = i + j;
      if (k >= total) break;
      const int c = (int)(k % C);
      const long long ns = k / C;          // b*HW + s
      const long long b = ns / HW;
      const long long s = ns - b * HW;
      const unsigned char v = x[(b * C + c) * HW + s];
      y[k] = DfdCvt<T>::from_f32(fmaf((float)v, inv_std[c], neg_mean_over_std[c]));
    }
  }
}

}  // namespace

at::Tensor normalize_uint8_nhwc(at::Tensor x, at::Tensor mean, at::Tensor std,
                                std::string dtype_s, bool channels_last) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kByte && x.dim() == 4,
              "normalize_uint8_nhwc: 4D uint8 CUDA tensor expected");
  TORCH_CHECK(x.is_contiguous(), "normalize_uint8_nhwc: NCHW-contiguous input expected");
  const int C = (int)x.size(1);
  const long long HW = (long long)x.size(2) * x.size(3);
  const long long total = (long long)x.size(0) * C * HW;

  at::ScalarType out_t = at::kFloat;
  if (dtype_s == "bfloat16") out_t = at::kBFloat16;
  else if (dtype_s == "float16") out_t = at::kHalf;
  else TORCH_CHECK(dtype_s == "float32", "unknown dtype ", dtype_s);

  auto y = at::empty(x.sizes(), x.options().dtype(out_t),
                     at::MemoryFormat::ChannelsLast);
  auto inv_std = (1.0 / std.to(at::kFloat)).contiguous();
  auto nmos = (-(mean.to(at::kFloat)) / std.to(at::kFloat)).contiguous();

  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const int grid = dfd_grid(total / 4 + 1, block);

  if (out_t == at::kBFloat16) {
    hipLaunchKernelGGL((normalize_u8_nhwc_kernel<__hip_bfloat16>), dim3(grid), dim3(block), 0,
                       stream, x.data_ptr<unsigned char>(), (__hip_bfloat16*)y.data_ptr(),
                       inv_std.data_ptr<float>(), nmos.data_ptr<float>(), total, C, HW);
  } else if (out_t == at::kHalf) {
    hipLaunchKernelGGL((normalize_u8_nhwc_kernel<__half>), dim3(grid), dim3(block), 0,
                       stream, x.data_ptr<unsigned char>(), (__half*)y.data_ptr(),
                       inv_std.data_ptr<float>(), nmos.data_ptr<float>(), total, C, HW);
  } else {
    hipLaunchKernelGGL((normalize_u8_nhwc_kernel<float>), dim3(grid), dim3(block), 0,
                       stream, x.data_ptr<unsigned char>(), (float*)y.data_ptr(),
                       inv_std.data_ptr<float>(), nmos.data_ptr<float>(), total, C, HW);
  }
  return y;
}
