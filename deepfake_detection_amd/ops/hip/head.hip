// Fused classifier head: logits = x @ W^T + b, label-smoothed softmax
// cross-entropy loss, and the fused backward (dx, dW, db) — SURVEY.md §2.6
// item 10 (reference dfd/timm/loss/cross_entropy.py:6-26 + the classifier
// Linear, efficientnet.py:300). The head is tiny (B x F x C, C=2 for
// deepfake), so the win is kernel-count (one fwd kernel instead of GEMM +
// log_softmax + nll + mean chains), not bandwidth.
//
// Shapes: x [B, F] fp32/bf16 pooled features; W [C, F]; b [C]; target [B]
// int64. C <= kMaxClasses (LDS logits per block). Loss = mean over B of
// smoothed NLL: (1-eps)*nll + eps*mean_c(-log p_c)  — exactly the
// reference LabelSmoothingCrossEntropy (eps=0 -> plain CE).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int kMaxClasses = 64;

template <typename T>
__global__ void head_fwd_kernel(
    const T* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ bias, const long long* __restrict__ target,
    float* __restrict__ logits,  // [B, C] (saved for metrics + backward)
    float* __restrict__ loss_sum,  // [1] accumulated
    int B, int F, int C, float smoothing) {
  extern __shared__ float lds[];  // [C] logits + scratch[16]
  const int b = blockIdx.x;
  if (b >= B) return;
  const int tid = threadIdx.x;
  float* scratch = lds + C;

  for (int c = 0; c < C; ++c) {
    float part = 0.f;
    for (int f = tid; f < F; f += blockDim.x)
      part += DfdCvt<T>::to_f32(x[(long long)b * F + f]) * w[(long long)c * F + f];
    const float tot = block_sum(part, scratch);
    if (tid == 0) lds[c] = tot + (bias ? bias[c] : 0.f);
    __syncthreads();
  }

  if (tid == 0) {
    float mx = lds[0];
    for (int c = 1; c < C; ++c) mx = fmaxf(mx, lds[c]);
    float se = 0.f;
    for (int c = 0; c < C; ++c) se += __expf(lds[c] - mx);
    const float lse = mx + __logf(se);
    float mean_nll = 0.f;
    for (int c = 0; c < C; ++c) {
      logits[(long long)b * C + c] = lds[c];
      mean_nll += lse - lds[c];
    }
    mean_nll /= (float)C;
    const int t = (int)target[b];
    const float nll = lse - lds[t];
    atomicAdd(loss_sum, ((1.f - smoothing) * nll + smoothing * mean_nll) / (float)B);
  }
}

// dlogits[b,c] = dloss/B * (softmax(logits)[b,c] - y_smooth[b,c])
__global__ void head_bwd_dlogits_kernel(
    const float* __restrict__ logits, const long long* __restrict__ target,
    const float* __restrict__ dloss, float* __restrict__ dlogits,
    int B, int C, float smoothing) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float mx = logits[(long long)b * C];
  for (int c = 1; c < C; ++c) mx = fmaxf(mx, logits[(long long)b * C + c]);
  float se = 0.f;
  for (int c = 0; c < C; ++c) se += __expf(logits[(long long)b * C + c] - mx);
  const float inv = 1.f / se;
  const int t = (int)target[b];
  const float g = dloss[0] / (float)B;
  const float off = smoothing / (float)C;
  for (int c = 0; c < C; ++c) {
    const float p = __expf(logits[(long long)b * C + c] - mx) * inv;
    const float y = (c == t ? 1.f - smoothing : 0.f) + off;
    dlogits[(long long)b * C + c] = g * (p - y);
  }
}

// dx[b,f] = sum_c dlogits[b,c] * w[c,f]
template <typename T>
__global__ void head_bwd_dx_kernel(const float* __restrict__ dlogits,
                                   const float* __restrict__ w, T* __restrict__ dx,
                                   long long total, int F, int C) {
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    const long long bb = i / F;
    const int f = (int)(i - bb * F);
    float v = 0.f;
    for (int c = 0; c < C; ++c)
      v += dlogits[bb * C + c] * w[(long long)c * F + f];
    dx[i] = DfdCvt<T>::from_f32(v);
  }
}

// dW[c,f] = sum_b dlogits[b,c] * x[b,f];  db[c] = sum_b dlogits[b,c]
template <typename T>
__global__ void head_bwd_dw_kernel(const float* __restrict__ dlogits,
                                   const T* __restrict__ x, float* __restrict__ dw,
                                   float* __restrict__ db, int B, int F, int C) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < (long long)C * F) {
    const int c = (int)(i / F);
    const int f = (int)(i - (long long)c * F);
    float v = 0.f;
    for (int b = 0; b < B; ++b)
      v += dlogits[(long long)b * C + c] * DfdCvt<T>::to_f32(x[(long long)b * F + f]);
    dw[i] = v;
  }
  if (db != nullptr && i < (long long)C) {
    float v = 0.f;
    for (int b = 0; b < B; ++b) v += dlogits[(long long)b * C + (int)i];
    db[i] = v;
  }
}

}  // namespace

// Returns {loss [scalar fp32], logits [B, C] fp32}.
std::vector<at::Tensor> head_ce_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                                    at::Tensor target, double smoothing) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2, "head: x must be [B, F] CUDA");
  TORCH_CHECK(w.scalar_type() == at::kFloat && bias.scalar_type() == at::kFloat,
              "head: fp32 classifier params required");
  TORCH_CHECK(target.scalar_type() == at::kLong, "head: int64 targets");
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  const int B = (int)x.size(0), F = (int)x.size(1), C = (int)w.size(0);
  TORCH_CHECK(C <= kMaxClasses, "head: too many classes for the fused head");
  auto opts_f = x.options().dtype(at::kFloat);
  auto logits = at::empty({B, C}, opts_f);
  auto loss = at::zeros({}, opts_f);
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int lds = (C + 16) * sizeof(float);
#define HEAD_FWD_T(T)                                                        \
  head_fwd_kernel<T><<<B, 256, lds, stream>>>(                               \
      (const T*)xc.data_ptr(), wc.data_ptr<float>(), bias.data_ptr<float>(), \
      (const long long*)target.data_ptr(), logits.data_ptr<float>(),         \
      loss.data_ptr<float>(), B, F, C, (float)smoothing)
  switch (x.scalar_type()) {
    case at::kBFloat16: HEAD_FWD_T(__hip_bfloat16); break;
    case at::kHalf: HEAD_FWD_T(__half); break;
    case at::kFloat: HEAD_FWD_T(float); break;
    default: TORCH_CHECK(false, "head: unsupported dtype");
  }
#undef HEAD_FWD_T
  return {loss, logits};
}

// Returns {dx [B, F] (x dtype), dw [C, F] fp32, db [C] fp32}.
std::vector<at::Tensor> head_ce_bwd(at::Tensor dloss, at::Tensor logits,
                                    at::Tensor x, at::Tensor w, at::Tensor target,
                                    double smoothing) {
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  const int B = (int)x.size(0), F = (int)x.size(1), C = (int)w.size(0);
  auto opts_f = x.options().dtype(at::kFloat);
  auto dlogits = at::empty({B, C}, opts_f);
  auto dx = at::empty_like(xc);
  auto dw = at::empty({C, F}, opts_f);
  auto db = at::empty({C}, opts_f);
  auto stream = at::hip::getCurrentHIPStream().stream();
  head_bwd_dlogits_kernel<<<dim3((B + 255) / 256), 256, 0, stream>>>(
      logits.data_ptr<float>(), (const long long*)target.data_ptr(),
      dloss.contiguous().data_ptr<float>(), dlogits.data_ptr<float>(), B, C,
      (float)smoothing);
  const long long total = (long long)B * F;
#define HEAD_BWD_T(T)                                                          \
  do {                                                                         \
    head_bwd_dx_kernel<T><<<dfd_grid(total, 256), 256, 0, stream>>>(           \
        dlogits.data_ptr<float>(), wc.data_ptr<float>(), (T*)dx.data_ptr(),    \
        total, F, C);                                                          \
    head_bwd_dw_kernel<T><<<dim3((unsigned)(((long long)C * F + 255) / 256)),  \
                            256, 0, stream>>>(                                 \
        dlogits.data_ptr<float>(), (const T*)xc.data_ptr(),                    \
        dw.data_ptr<float>(), db.data_ptr<float>(), B, F, C);                  \
  } while (0)
  switch (x.scalar_type()) {
    case at::kBFloat16: HEAD_BWD_T(__hip_bfloat16); break;
    case at::kHalf: HEAD_BWD_T(__half); break;
    case at::kFloat: HEAD_BWD_T(float); break;
    default: TORCH_CHECK(false, "head: unsupported dtype");
  }
#undef HEAD_BWD_T
  return {dx, dw, db};
}
