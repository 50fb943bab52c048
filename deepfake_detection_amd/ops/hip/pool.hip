// Global average pool NHWC fwd/bwd (head pooling).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// one block per sample n; threads stride channels, loop spatial rows;
// fp32 accumulate, output dtype = input dtype, shape (N, C)
template <typename T>
__global__ void gap_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                               int C, long long HW) {
  const long long n = blockIdx.x;
  const T* xn = x + n * HW * C;
  const float inv = 1.f / (float)HW;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float acc = 0.f;
    for (long long s = 0; s < HW; ++s) acc += DfdCvt<T>::to_f32(xn[s * C + c]);
    y[n * C + c] = DfdCvt<T>::from_f32(acc * inv);
  }
}

template <typename T>
__global__ void gap_bwd_kernel(const T* __restrict__ dy, T* __restrict__ dx,
                               long long total, int C, long long HW) {
  const long long idx0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const float inv = 1.f / (float)HW;
  for (long long k = idx0; k < total; k += stride) {
    const int c = (int)(k % C);
    const long long n = k / (HW * C);
    dx[k] = DfdCvt<T>::from_f32(DfdCvt<T>::to_f32(dy[n * C + c]) * inv);
  }
}

}  // namespace

at::Tensor global_avg_pool_fwd(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "gap_fwd: 4D CUDA tensor expected");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "gap_fwd: channels_last input required");
  const long long N = x.size(0);
  const int C = (int)x.size(1);
  const long long HW = (long long)x.size(2) * x.size(3);
  auto y = at::empty({N, (long long)C}, x.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
#define LAUNCH(T)                                                             \
  hipLaunchKernelGGL((gap_fwd_kernel<T>), dim3((unsigned)N), dim3(block), 0,  \
                     stream, (const T*)x.data_ptr(), (T*)y.data_ptr(), C, HW)
  if (x.scalar_type() == at::kBFloat16) LAUNCH(__hip_bfloat16);
  else if (x.scalar_type() == at::kHalf) LAUNCH(__half);
  else if (x.scalar_type() == at::kFloat) LAUNCH(float);
  else TORCH_CHECK(false, "gap_fwd: unsupported dtype");
#undef LAUNCH
  return y;
}

at::Tensor global_avg_pool_bwd(at::Tensor dy, long long N, long long C,
                               long long H, long long W) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 2, "gap_bwd: 2D CUDA tensor expected");
  dy = dy.contiguous();
  auto dx = at::empty({N, C, H, W}, dy.options(), at::MemoryFormat::ChannelsLast);
  const long long HW = H * W;
  const long long total = N * C * HW;
  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const int grid = dfd_grid(total, block);
#define LAUNCH(T)                                                              \
  hipLaunchKernelGGL((gap_bwd_kernel<T>), dim3(grid), dim3(block), 0, stream,  \
                     (const T*)dy.data_ptr(), (T*)dx.data_ptr(), total, (int)C, HW)
  if (dy.scalar_type() == at::kBFloat16) LAUNCH(__hip_bfloat16);
  else if (dy.scalar_type() == at::kHalf) LAUNCH(__half);
  else if (dy.scalar_type() == at::kFloat) LAUNCH(float);
  else TORCH_CHECK(false, "gap_bwd: unsupported dtype");
#undef LAUNCH
  return dx;
}
