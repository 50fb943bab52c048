// Fused multi-tensor optimizer / EMA kernels.
//
// One kernel launch updates every parameter tensor of a group: the host
// packs {pointers, sizes} into a chunk table (int64 tensor shipped to the
// device once per step), each block grabs one chunk and grid-strides it.
// Replaces the reference's ~6 eager kernels per tensor for RMSpropTF
// (reference dfd/timm/optim/rmsprop_tf.py:86-120), AdamW, and the EMA loop
// (reference dfd/timm/utils.py:329-340).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

#include <cstring>

#include "common.h"

namespace {

// Elements per chunk: small enough that a 62M-param model yields thousands
// of blocks (the first cut used 1M-element chunks -> ~220 blocks -> latency
// bound at 2.7 ms/step; roofline for the update is ~0.3 ms).
constexpr long long kChunk = 1 << 15;

// chunk table layout (int64 per chunk): [p, g, m1, m2, numel]
struct ChunkTable {
  at::Tensor dev;  // [nchunks, 5] int64 on device
  int nchunks;
};

// Capture-safe table fill: the values travel as KERNEL ARGUMENTS (embedded
// in the graph node), so building a table inside an active hipGraph capture
// needs no host allocation and no memcpy — both of which invalidate a
// capture (hipHostMalloc/pageable H2D).
struct TableSlice {
  long long v[480];
};

__global__ void fill_table_kernel(long long* __restrict__ dst, TableSlice s, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dst[i] = s.v[i];
}

ChunkTable build_chunks(const std::vector<at::Tensor>& a,
                        const std::vector<at::Tensor>& b,
                        const std::vector<at::Tensor>& c,
                        const std::vector<at::Tensor>& d,
                        const at::TensorOptions& opts) {
  std::vector<long long> rows;
  const size_t n = a.size();
  for (size_t i = 0; i < n; ++i) {
    const long long numel = a[i].numel();
    for (long long off = 0; off < numel; off += kChunk) {
      const long long len = std::min(kChunk, numel - off);
      rows.push_back((long long)a[i].data_ptr() + off * a[i].element_size());
      rows.push_back(b.empty() ? 0 : (long long)b[i].data_ptr() + off * b[i].element_size());
      rows.push_back(c.empty() ? 0 : (long long)c[i].data_ptr() + off * c[i].element_size());
      rows.push_back(d.empty() ? 0 : (long long)d[i].data_ptr() + off * d[i].element_size());
      rows.push_back(len);
    }
  }
  const int nchunks = (int)(rows.size() / 5);

  // Tensor addresses are stable across steps in steady state (params always;
  // grads under gradient_as_bucket_view or the caching allocator), so cache
  // the device table keyed on its full contents: skips a host-alloc + H2D
  // per step and keeps optimizer steps hipGraph-capturable.
  struct Cache {
    std::vector<long long> key;
    at::Tensor host;  // pinned staging — must outlive any captured memcpy node
    at::Tensor dev;
  };
  static thread_local std::vector<Cache> cache;
  for (auto& e : cache) {
    if (e.key == rows && e.dev.device() == opts.device()) {
      ChunkTable t;
      t.dev = e.dev;
      t.nchunks = nchunks;
      return t;
    }
  }
  ChunkTable t;
  t.nchunks = nchunks;
  hipStreamCaptureStatus cap = hipStreamCaptureStatusNone;
  auto stream = at::hip::getCurrentHIPStream().stream();
  (void)hipStreamIsCapturing(stream, &cap);
  at::Tensor host;  // kept alive by the cache when the pinned path is used
  if (cap != hipStreamCaptureStatusNone) {
    // Inside capture (grads live at graph-pool addresses -> cache miss):
    // device alloc is legal (graph pool); ship the values as kernel args.
    t.dev = at::empty({(long long)nchunks, 5},
                      at::TensorOptions().dtype(at::kLong).device(opts.device()));
    long long* dst = (long long*)t.dev.data_ptr();
    const long long total = (long long)rows.size();
    for (long long off = 0; off < total; off += 480) {
      TableSlice s;
      const int n = (int)std::min<long long>(480, total - off);
      std::memcpy(s.v, rows.data() + off, n * sizeof(long long));
      fill_table_kernel<<<dim3((n + 255) / 256), 256, 0, stream>>>(dst + off, s, n);
    }
  } else {
    // Eager path: pinned staging + one async H2D.
    host = at::empty({(long long)nchunks, 5},
                     at::TensorOptions().dtype(at::kLong).pinned_memory(true));
    std::memcpy(host.data_ptr(), rows.data(), rows.size() * sizeof(long long));
    t.dev = host.to(opts.device(), /*non_blocking=*/true);
  }
  if (cache.size() > 8) cache.clear();  // bound: a few optimizers/EMA per process
  cache.push_back({std::move(rows), host, t.dev});
  return t;
}

DFD_DEV void rmsprop_tf_update(float& pv, float gv, float& s, float& b, float lr,
                               float oma, float eps, float momentum,
                               float weight_decay, bool decoupled, bool lr_in_mom) {
  if (weight_decay != 0.f) {
    if (decoupled) pv -= weight_decay * pv;
    else gv += weight_decay * pv;
  }
  s += oma * (gv * gv - s);          // TF order of ops
  const float avg = sqrtf(s + eps);  // eps inside sqrt
  if (momentum > 0.f) {
    if (lr_in_mom) {
      b = b * momentum + lr * gv / avg;  // LR inside the buffer
      pv -= b;
    } else {
      b = b * momentum + gv / avg;
      pv -= lr * b;
    }
  } else {
    pv -= lr * gv / avg;
  }
}

__global__ void rmsprop_tf_kernel(const long long* __restrict__ table, int nchunks,
                                  float lr, float alpha, float eps, float momentum,
                                  float weight_decay, bool decoupled, bool lr_in_mom) {
  const float oma = 1.f - alpha;
  for (int ch = blockIdx.x; ch < nchunks; ch += gridDim.x) {
    const long long* row = table + (long long)ch * 5;
    float* p = (float*)row[0];
    const float* g = (const float*)row[1];
    float* sa = (float*)row[2];
    float* buf = (float*)row[3];
    const long long nelem = row[4];
    const long long nv = nelem / 4;
    const bool has_mom = momentum > 0.f;
    for (long long i = threadIdx.x; i < nv; i += blockDim.x) {
      float4 pv = ((float4*)p)[i];
      const float4 gv = ((const float4*)g)[i];
      float4 sv = ((float4*)sa)[i];
      float4 bv = has_mom ? ((float4*)buf)[i] : float4{0, 0, 0, 0};
      rmsprop_tf_update(pv.x, gv.x, sv.x, bv.x, lr, oma, eps, momentum, weight_decay, decoupled, lr_in_mom);
      rmsprop_tf_update(pv.y, gv.y, sv.y, bv.y, lr, oma, eps, momentum, weight_decay, decoupled, lr_in_mom);
      rmsprop_tf_update(pv.z, gv.z, sv.z, bv.z, lr, oma, eps, momentum, weight_decay, decoupled, lr_in_mom);
      rmsprop_tf_update(pv.w, gv.w, sv.w, bv.w, lr, oma, eps, momentum, weight_decay, decoupled, lr_in_mom);
      ((float4*)p)[i] = pv;
      ((float4*)sa)[i] = sv;
      if (has_mom) ((float4*)buf)[i] = bv;
    }
    for (long long i = nv * 4 + threadIdx.x; i < nelem; i += blockDim.x) {
      float pv = p[i], s = sa[i], b = has_mom ? buf[i] : 0.f;
      rmsprop_tf_update(pv, g[i], s, b, lr, oma, eps, momentum, weight_decay, decoupled, lr_in_mom);
      p[i] = pv;
      sa[i] = s;
      if (has_mom) buf[i] = b;
    }
  }
}

__global__ void adamw_kernel(const long long* __restrict__ table, int nchunks,
                             float lr, float beta1, float beta2, float eps,
                             float weight_decay, float bias_c1, float inv_sqrt_bias_c2) {
  for (int ch = blockIdx.x; ch < nchunks; ch += gridDim.x) {
    const long long* row = table + (long long)ch * 5;
    float* p = (float*)row[0];
    const float* g = (const float*)row[1];
    float* m1 = (float*)row[2];
    float* m2 = (float*)row[3];
    const long long nelem = row[4];
    const float step_size = lr / bias_c1;
    const long long nv = nelem / 4;
    for (long long i = threadIdx.x; i < nv; i += blockDim.x) {
      float4 pv = ((float4*)p)[i];
      const float4 gv = ((const float4*)g)[i];
      float4 av = ((float4*)m1)[i];
      float4 vv = ((float4*)m2)[i];
      float* pp = &pv.x;
      const float* gp = &gv.x;
      float* ap = &av.x;
      float* vp = &vv.x;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float pd = pp[j] * (1.f - lr * weight_decay);
        const float a = ap[j] = beta1 * ap[j] + (1.f - beta1) * gp[j];
        const float v = vp[j] = beta2 * vp[j] + (1.f - beta2) * gp[j] * gp[j];
        pp[j] = pd - step_size * a / (sqrtf(v) * inv_sqrt_bias_c2 + eps);
      }
      ((float4*)p)[i] = pv;
      ((float4*)m1)[i] = av;
      ((float4*)m2)[i] = vv;
    }
    for (long long i = nv * 4 + threadIdx.x; i < nelem; i += blockDim.x) {
      float pv = p[i] * (1.f - lr * weight_decay);
      const float gv = g[i];
      const float a = m1[i] = beta1 * m1[i] + (1.f - beta1) * gv;
      const float v = m2[i] = beta2 * m2[i] + (1.f - beta2) * gv * gv;
      const float denom = sqrtf(v) * inv_sqrt_bias_c2 + eps;
      p[i] = pv - step_size * a / denom;
    }
  }
}

template <typename T>
__global__ void ema_kernel(const long long* __restrict__ table, int nchunks, float decay) {
  for (int ch = blockIdx.x; ch < nchunks; ch += gridDim.x) {
    const long long* row = table + (long long)ch * 5;
    T* e = (T*)row[0];
    const T* m = (const T*)row[1];
    const long long nelem = row[4];
    for (long long i = threadIdx.x; i < nelem; i += blockDim.x) {
      const float ev = DfdCvt<T>::to_f32(e[i]);
      const float mv = DfdCvt<T>::to_f32(m[i]);
      e[i] = DfdCvt<T>::from_f32(ev * decay + (1.f - decay) * mv);
    }
  }
}

}  // namespace

void rmsprop_tf_multi_tensor(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                             std::vector<at::Tensor> square_avgs,
                             std::vector<at::Tensor> momentum_buffers,
                             double lr, double alpha, double eps, double momentum,
                             double weight_decay, bool decoupled_decay, bool lr_in_momentum) {
  TORCH_CHECK(!params.empty());
  for (auto& p : params)
    TORCH_CHECK(p.scalar_type() == at::kFloat && p.is_non_overlapping_and_dense(),
                "rmsprop_tf_multi_tensor: dense fp32 params required");
  auto table = build_chunks(params, grads, square_avgs, momentum_buffers, params[0].options());
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsprop_tf_kernel, dim3(std::min(table.nchunks, kMaxGrid)), dim3(256), 0,
                     stream, (const long long*)table.dev.data_ptr<int64_t>(), table.nchunks,
                     (float)lr, (float)alpha, (float)eps, (float)momentum,
                     (float)weight_decay, decoupled_decay, lr_in_momentum);
}

void adamw_multi_tensor(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                        std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
                        long long step, double lr, double beta1, double beta2,
                        double eps, double weight_decay) {
  TORCH_CHECK(!params.empty());
  for (auto& p : params)
    TORCH_CHECK(p.scalar_type() == at::kFloat && p.is_non_overlapping_and_dense(),
                "adamw_multi_tensor: dense fp32 params required");
  auto table = build_chunks(params, grads, exp_avgs, exp_avg_sqs, params[0].options());
  auto stream = at::cuda::getCurrentHIPStream();
  const float bias_c1 = 1.f - powf((float)beta1, (float)step);
  const float bias_c2 = 1.f - powf((float)beta2, (float)step);
  hipLaunchKernelGGL(adamw_kernel, dim3(std::min(table.nchunks, kMaxGrid)), dim3(256), 0,
                     stream, (const long long*)table.dev.data_ptr<int64_t>(), table.nchunks,
                     (float)lr, (float)beta1, (float)beta2, (float)eps,
                     (float)weight_decay, bias_c1, 1.f / sqrtf(bias_c2));
}

void ema_multi_tensor(std::vector<at::Tensor> ema_params, std::vector<at::Tensor> model_params,
                      double decay) {
  TORCH_CHECK(!ema_params.empty());
  const auto st = ema_params[0].scalar_type();
  for (size_t i = 0; i < ema_params.size(); ++i) {
    TORCH_CHECK(ema_params[i].scalar_type() == st && model_params[i].scalar_type() == st);
  }
  std::vector<at::Tensor> empty;
  auto table = build_chunks(ema_params, model_params, empty, empty, ema_params[0].options());
  auto stream = at::cuda::getCurrentHIPStream();
  const dim3 grid(std::min(table.nchunks, kMaxGrid));
  if (st == at::kFloat) {
    hipLaunchKernelGGL(ema_kernel<float>, grid, dim3(256), 0, stream,
                       (const long long*)table.dev.data_ptr<int64_t>(), table.nchunks, (float)decay);
  } else if (st == at::kBFloat16) {
    hipLaunchKernelGGL(ema_kernel<__hip_bfloat16>, grid, dim3(256), 0, stream,
                       (const long long*)table.dev.data_ptr<int64_t>(), table.nchunks, (float)decay);
  } else if (st == at::kHalf) {
    hipLaunchKernelGGL(ema_kernel<__half>, grid, dim3(256), 0, stream,
                       (const long long*)table.dev.data_ptr<int64_t>(), table.nchunks, (float)decay);
  } else {
    TORCH_CHECK(false, "ema_multi_tensor: unsupported dtype");
  }
}
