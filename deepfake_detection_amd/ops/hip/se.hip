// Fused squeeze-excite chain, NHWC, gfx950.
//
// Replaces the reference's 6-kernel chain per SE block
// (reference efficientnet_blocks.py:104-110). Forward:
//   1. se_pool_kernel    — s[n,c] = mean_hw x  (chunked over HW, fp32 atomics)
//   2. se_fc_gate_kernel — 1x1 reduce GEMV -> act -> 1x1 expand GEMV -> sigmoid
//   3. gate_apply_kernel — y = x * g (vectorized broadcast over HW)
// Backward:
//   4. se_bwd_reduce_kernel — dx = dy*g and dg[n,c] = sum_hw dy*x (chunked)
//   5. (python) dense chain: tiny (N,C)x(C,Cr) GEMMs in torch/rocBLAS
//   6. se_bwd_add_pool_kernel — dx += ds/HW (vectorized broadcast)
//
// Grids are sized to fill 256 CUs even for small-C layers: threads map to
// (channel-slot, row-group) with a power-of-two channel-slot count chosen
// from C, and HW is chunked across blockIdx.z. First version launched one
// block per sample and serially pooled HW — 17% of step time
// (profiles/r01: se_gate_fwd 1.2 ms avg); this layout is bandwidth-bound.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T, int N>
struct alignas(sizeof(T) * N) SVec {
  T v[N];
};

template <typename T, int N>
DFD_DEV SVec<T, N> svload(const T* p) {
  return *reinterpret_cast<const SVec<T, N>*>(p);
}

template <typename T, int N>
DFD_DEV void svstore(T* p, const SVec<T, N>& x) {
  *reinterpret_cast<SVec<T, N>*>(p) = x;
}

// ---------------------------------------------------------------------------
// pool: s[n,c] += sum over a chunk of HW rows (host zeroes s, scales by 1/HW)
// block: cpb channel-slots (power of two) x nrg row-groups; VEC channels/slot
// grid: (N, ceil(cv/cpb), hw_chunks)
// ---------------------------------------------------------------------------
template <typename T, int VEC>
__global__ void se_pool_kernel(const T* __restrict__ x, float* __restrict__ s, int C,
                               long long HW, int cpb, int rows_per_chunk) {
  extern __shared__ float lds[];  // [blockDim.x * VEC]
  const int slot = threadIdx.x % cpb;
  const int rg = threadIdx.x / cpb;
  const int nrg = blockDim.x / cpb;
  const int cv = C / VEC;
  const int cvec = blockIdx.y * cpb + slot;
  const long long n = blockIdx.x;
  const bool active = cvec < cv && rg < nrg;  // tail threads must not stream
  const int c = cvec * VEC;

  float acc[VEC];
#pragma unroll
  for (int i = 0; i < VEC; ++i) acc[i] = 0.f;

  if (active) {
    const T* xn = x + n * HW * C + c;
    const long long r0 = (long long)blockIdx.z * rows_per_chunk;
    const long long r1 = min(r0 + rows_per_chunk, HW);
    long long r = r0 + rg;
    for (; r + 3 * (long long)nrg < r1; r += 4 * (long long)nrg) {
      SVec<T, VEC> xv[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) xv[u] = svload<T, VEC>(xn + (r + u * (long long)nrg) * C);
#pragma unroll
      for (int u = 0; u < 4; ++u)
#pragma unroll
        for (int i = 0; i < VEC; ++i) acc[i] += DfdCvt<T>::to_f32(xv[u].v[i]);
    }
    for (; r < r1; r += nrg) {
      const SVec<T, VEC> xv = svload<T, VEC>(xn + r * C);
#pragma unroll
      for (int i = 0; i < VEC; ++i) acc[i] += DfdCvt<T>::to_f32(xv.v[i]);
    }
  }

  // cross-row-group tree reduction in LDS (non-pow2 nrg: fold excess first)
  float* my = lds + (size_t)(rg * cpb + slot) * VEC;
  int p2 = 1;
  while (p2 * 2 <= nrg) p2 *= 2;
  const bool in_block = rg < nrg;
  if (in_block) {
#pragma unroll
    for (int i = 0; i < VEC; ++i) my[i] = acc[i];
  }
  __syncthreads();
  if (in_block && rg >= p2) {
    float* dst = lds + (size_t)((rg - p2) * cpb + slot) * VEC;
#pragma unroll
    for (int i = 0; i < VEC; ++i) dst[i] += my[i];
  }
  __syncthreads();
  for (int step = p2 >> 1; step > 0; step >>= 1) {
    if (rg < step) {
      const float* other = lds + ((size_t)((rg + step) * cpb) + slot) * VEC;
#pragma unroll
      for (int i = 0; i < VEC; ++i) my[i] += other[i];
    }
    __syncthreads();
  }
  if (rg == 0 && cvec < cv) {
#pragma unroll
    for (int i = 0; i < VEC; ++i) atomicAdd(s + n * C + c + i, my[i]);
  }
}

// ---------------------------------------------------------------------------
// fc gate: z1 = w1 . s + b1; r = act(z1); g = sigmoid(w2 . r + b2)
// grid = N blocks (the dense work is tiny: C*Cr MACs per sample)
// ---------------------------------------------------------------------------
template <Act ACT>
__global__ void se_fc_gate_kernel(const float* __restrict__ s, const float* __restrict__ w1,
                                  const float* __restrict__ b1, const float* __restrict__ w2,
                                  const float* __restrict__ b2, float* __restrict__ z1_out,
                                  float* __restrict__ r_out, float* __restrict__ g_out,
                                  int C, int Cr) {
  extern __shared__ float smem[];
  float* s_lds = smem;       // [C]
  float* r_lds = smem + C;   // [Cr]
  const long long n = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wid = tid / kWave;
  const int nw = blockDim.x / kWave;

  for (int c = tid; c < C; c += blockDim.x) s_lds[c] = s[n * C + c];
  __syncthreads();

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

  for (int c = tid; c < C; c += blockDim.x) {
    const float* wrow = w2 + (long long)c * Cr;
    float d = b2 ? b2[c] : 0.f;
    for (int j = 0; j < Cr; ++j) d += wrow[j] * r_lds[j];
    g_out[n * C + c] = 1.f / (1.f + __expf(-d));
  }
}

// ---------------------------------------------------------------------------
// y = x * g (broadcast over HW), vectorized over channels
// ---------------------------------------------------------------------------
template <typename T, int VEC>
__global__ void gate_apply_kernel(const T* __restrict__ x, const float* __restrict__ g,
                                  T* __restrict__ y, long long NHW, int C, long long HW) {
  const int cv = C / VEC;
  const long long total = NHW * cv;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(idx % cv) * VEC;
    const long long p = idx / cv;        // flat (n*HW + hw)
    const long long n = p / HW;
    const SVec<T, VEC> xv = svload<T, VEC>(x + p * C + c);
    SVec<T, VEC> yv;
#pragma unroll
    for (int i = 0; i < VEC; ++i)
      yv.v[i] = DfdCvt<T>::from_f32(DfdCvt<T>::to_f32(xv.v[i]) * g[n * C + c + i]);
    svstore<T, VEC>(y + p * C + c, yv);
  }
}

// ---------------------------------------------------------------------------
// backward pass 1: dx = dy * g  and  dg[n,c] = sum_hw dy * x (chunked+atomic)
// same block layout as se_pool_kernel
// ---------------------------------------------------------------------------
template <typename T, int VEC>
__global__ void se_bwd_reduce_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                     const float* __restrict__ g, T* __restrict__ dx,
                                     float* __restrict__ dg, int C, long long HW,
                                     int cpb, int rows_per_chunk) {
  extern __shared__ float lds[];
  const int slot = threadIdx.x % cpb;
  const int rg = threadIdx.x / cpb;
  const int nrg = blockDim.x / cpb;
  const int cv = C / VEC;
  const int cvec = blockIdx.y * cpb + slot;
  const long long n = blockIdx.x;
  const bool active = cvec < cv && rg < nrg;
  const int c = cvec * VEC;

  float acc[VEC], gv[VEC];
#pragma unroll
  for (int i = 0; i < VEC; ++i) acc[i] = 0.f;

  if (active) {
#pragma unroll
    for (int i = 0; i < VEC; ++i) gv[i] = g[n * C + c + i];
    const T* xn = x + n * HW * C + c;
    const T* dn = dy + n * HW * C + c;
    T* dxn = dx + n * HW * C + c;
    const long long r0 = (long long)blockIdx.z * rows_per_chunk;
    const long long r1 = min(r0 + rows_per_chunk, HW);
    long long r = r0 + rg;
    // 4-row batches: 8 independent loads in flight per iteration
    for (; r + 3 * (long long)nrg < r1; r += 4 * (long long)nrg) {
      SVec<T, VEC> xv[4], dv[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        xv[u] = svload<T, VEC>(xn + (r + u * (long long)nrg) * C);
        dv[u] = svload<T, VEC>(dn + (r + u * (long long)nrg) * C);
      }
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        SVec<T, VEC> ov;
#pragma unroll
        for (int i = 0; i < VEC; ++i) {
          const float d = DfdCvt<T>::to_f32(dv[u].v[i]);
          acc[i] += d * DfdCvt<T>::to_f32(xv[u].v[i]);
          ov.v[i] = DfdCvt<T>::from_f32(d * gv[i]);
        }
        svstore<T, VEC>(dxn + (r + u * (long long)nrg) * C, ov);
      }
    }
    for (; r < r1; r += nrg) {
      const SVec<T, VEC> xv = svload<T, VEC>(xn + r * C);
      const SVec<T, VEC> dv = svload<T, VEC>(dn + r * C);
      SVec<T, VEC> ov;
#pragma unroll
      for (int i = 0; i < VEC; ++i) {
        const float d = DfdCvt<T>::to_f32(dv.v[i]);
        acc[i] += d * DfdCvt<T>::to_f32(xv.v[i]);
        ov.v[i] = DfdCvt<T>::from_f32(d * gv[i]);
      }
      svstore<T, VEC>(dxn + r * C, ov);
    }
  }

  float* my = lds + (size_t)(rg * cpb + slot) * VEC;
  int p2 = 1;
  while (p2 * 2 <= nrg) p2 *= 2;
  const bool in_block = rg < nrg;
  if (in_block) {
#pragma unroll
    for (int i = 0; i < VEC; ++i) my[i] = acc[i];
  }
  __syncthreads();
  if (in_block && rg >= p2) {
    float* dst = lds + (size_t)((rg - p2) * cpb + slot) * VEC;
#pragma unroll
    for (int i = 0; i < VEC; ++i) dst[i] += my[i];
  }
  __syncthreads();
  for (int step = p2 >> 1; step > 0; step >>= 1) {
    if (rg < step) {
      const float* other = lds + ((size_t)((rg + step) * cpb) + slot) * VEC;
#pragma unroll
      for (int i = 0; i < VEC; ++i) my[i] += other[i];
    }
    __syncthreads();
  }
  if (rg == 0 && cvec < cv) {
#pragma unroll
    for (int i = 0; i < VEC; ++i) atomicAdd(dg + n * C + c + i, my[i]);
  }
}

// backward pass 2: dx += ds[n,c]/HW, vectorized
template <typename T, int VEC>
__global__ void se_bwd_add_pool_kernel(T* __restrict__ dx, const float* __restrict__ ds,
                                       long long NHW, int C, long long HW, float invHW) {
  const int cv = C / VEC;
  const long long total = NHW * cv;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(idx % cv) * VEC;
    const long long p = idx / cv;
    const long long n = p / HW;
    SVec<T, VEC> v = svload<T, VEC>(dx + p * C + c);
#pragma unroll
    for (int i = 0; i < VEC; ++i)
      v.v[i] = DfdCvt<T>::from_f32(DfdCvt<T>::to_f32(v.v[i]) + ds[n * C + c + i] * invHW);
    svstore<T, VEC>(dx + p * C + c, v);
  }
}

// ---------------------------------------------------------------------------
// host helpers
// ---------------------------------------------------------------------------
int se_pick_vec(long long c, int elem_size) {
  const int max_vec = elem_size == 4 ? 4 : 8;
  for (int v = max_vec; v > 1; v >>= 1)
    if (c % v == 0) return v;
  return 1;
}

// channel slots per block: exact (non-pow2) balanced tiles — rounding up to
// a power of two idled up to 44% of a block's load bandwidth (C=336: 42 of
// 64 slots active)
int pick_cpb(int cv, int* ctiles_out) {
  const int ctiles = (cv + 63) / 64;
  *ctiles_out = ctiles;
  return (cv + ctiles - 1) / ctiles;
}

struct ChunkPlan {
  int cpb, ctiles, chunks, rows_per_chunk;
};

ChunkPlan plan_chunks(long long N, int cv, long long HW) {
  ChunkPlan p;
  p.cpb = pick_cpb(cv, &p.ctiles);
  const long long base = N * p.ctiles;
  long long want = (2048 + base - 1) / base;  // chunks to reach ~2048 blocks
  const int nrg = 256 / p.cpb;
  // keep ≥~16 row-iterations per thread to amortize the per-block atomics
  long long by_iters = HW / ((long long)nrg * 16);
  if (want > by_iters) want = by_iters;
  long long max_chunks = (HW + nrg - 1) / nrg;
  if (want > max_chunks) want = max_chunks;
  if (want < 1) want = 1;
  p.chunks = (int)want;
  p.rows_per_chunk = (int)((HW + p.chunks - 1) / p.chunks);
  return p;
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

  auto w1c = w1.to(at::kFloat).contiguous();
  auto w2c = w2.to(at::kFloat).contiguous();
  auto b1c = b1.defined() ? b1.to(at::kFloat).contiguous() : at::Tensor();
  auto b2c = b2.defined() ? b2.to(at::kFloat).contiguous() : at::Tensor();

  auto opts_f = x.options().dtype(at::kFloat);
  auto s = at::zeros({N, (long long)C}, opts_f);
  auto z1 = at::empty({N, (long long)Cr}, opts_f);
  auto r = at::empty({N, (long long)Cr}, opts_f);
  auto g = at::empty({N, (long long)C}, opts_f);
  auto y = at::empty_like(x);

  auto stream = at::hip::getCurrentHIPStream().stream();
  const int vec = se_pick_vec(C, (int)x.element_size());
  const auto plan = plan_chunks(N, C / vec, HW);
  dim3 pool_grid((unsigned)N, plan.ctiles, plan.chunks);
  const int pool_lds = 256 * vec * sizeof(float);

#define LAUNCH_POOL(T, V)                                                       \
  se_pool_kernel<T, V><<<pool_grid, 256, pool_lds, stream>>>(                   \
      (const T*)x.data_ptr(), s.data_ptr<float>(), C, HW, plan.cpb,        \
      plan.rows_per_chunk)
#define POOL_VEC(T)                                                             \
  switch (vec) {                                                                \
    case 8: LAUNCH_POOL(T, 8); break;                                           \
    case 4: LAUNCH_POOL(T, 4); break;                                           \
    case 2: LAUNCH_POOL(T, 2); break;                                           \
    default: LAUNCH_POOL(T, 1); break;                                          \
  }
  if (x.scalar_type() == at::kBFloat16) { POOL_VEC(__hip_bfloat16) }
  else if (x.scalar_type() == at::kHalf) { POOL_VEC(__half) }
  else if (x.scalar_type() == at::kFloat) { POOL_VEC(float) }
  else TORCH_CHECK(false, "se_fwd: unsupported dtype");
#undef POOL_VEC
#undef LAUNCH_POOL
  s.mul_(1.0 / (double)HW);

  const int fc_lds = (C + Cr) * sizeof(float);
  if (act_s == "relu")
    se_fc_gate_kernel<Act::kRelu><<<dim3((unsigned)N), 256, fc_lds, stream>>>(
        s.data_ptr<float>(), w1c.data_ptr<float>(),
        b1c.defined() ? b1c.data_ptr<float>() : nullptr, w2c.data_ptr<float>(),
        b2c.defined() ? b2c.data_ptr<float>() : nullptr, z1.data_ptr<float>(),
        r.data_ptr<float>(), g.data_ptr<float>(), C, Cr);
  else
    se_fc_gate_kernel<Act::kSilu><<<dim3((unsigned)N), 256, fc_lds, stream>>>(
        s.data_ptr<float>(), w1c.data_ptr<float>(),
        b1c.defined() ? b1c.data_ptr<float>() : nullptr, w2c.data_ptr<float>(),
        b2c.defined() ? b2c.data_ptr<float>() : nullptr, z1.data_ptr<float>(),
        r.data_ptr<float>(), g.data_ptr<float>(), C, Cr);

  const long long NHW = N * HW;
#define LAUNCH_APPLY(T, V)                                                      \
  gate_apply_kernel<T, V><<<dfd_grid(NHW*(C / V), 256), 256, 0, stream>>>(      \
      (const T*)x.data_ptr(), g.data_ptr<float>(), (T*)y.data_ptr(), NHW, C, HW)
#define APPLY_VEC(T)                                                            \
  switch (vec) {                                                                \
    case 8: LAUNCH_APPLY(T, 8); break;                                          \
    case 4: LAUNCH_APPLY(T, 4); break;                                          \
    case 2: LAUNCH_APPLY(T, 2); break;                                          \
    default: LAUNCH_APPLY(T, 1); break;                                         \
  }
  if (x.scalar_type() == at::kBFloat16) { APPLY_VEC(__hip_bfloat16) }
  else if (x.scalar_type() == at::kHalf) { APPLY_VEC(__half) }
  else { APPLY_VEC(float) }
#undef APPLY_VEC
#undef LAUNCH_APPLY

  return {y, s, z1, r, g};
}

// pass 1 of backward: returns {dx (gate-direct part), dg}
std::vector<at::Tensor> se_bwd_reduce(at::Tensor dy, at::Tensor x, at::Tensor g) {
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  const long long N = x.size(0);
  const int C = (int)x.size(1);
  const long long HW = (long long)x.size(2) * x.size(3);
  auto dx = at::empty_like(x);
  auto dg = at::zeros({N, (long long)C}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int vec = se_pick_vec(C, (int)x.element_size());
  const auto plan = plan_chunks(N, C / vec, HW);
  dim3 grid((unsigned)N, plan.ctiles, plan.chunks);
  const int lds = 256 * vec * sizeof(float);

#define LAUNCH(T, V)                                                            \
  se_bwd_reduce_kernel<T, V><<<grid, 256, lds, stream>>>(                       \
      (const T*)dy.data_ptr(), (const T*)x.data_ptr(), g.data_ptr<float>(),     \
      (T*)dx.data_ptr(), dg.data_ptr<float>(), C, HW, plan.cpb,            \
      plan.rows_per_chunk)
#define LV(T)                                                                   \
  switch (vec) {                                                                \
    case 8: LAUNCH(T, 8); break;                                                \
    case 4: LAUNCH(T, 4); break;                                                \
    case 2: LAUNCH(T, 2); break;                                                \
    default: LAUNCH(T, 1); break;                                               \
  }
  if (x.scalar_type() == at::kBFloat16) { LV(__hip_bfloat16) }
  else if (x.scalar_type() == at::kHalf) { LV(__half) }
  else if (x.scalar_type() == at::kFloat) { LV(float) }
  else TORCH_CHECK(false, "se_bwd_reduce: unsupported dtype");
#undef LV
#undef LAUNCH
  return {dx, dg};
}

// pass 2: dx += ds/HW in place
void se_bwd_add_pool(at::Tensor dx, at::Tensor ds) {
  const long long N = dx.size(0);
  const int C = (int)dx.size(1);
  const long long HW = (long long)dx.size(2) * dx.size(3);
  const long long NHW = N * HW;
  auto dsc = ds.contiguous();
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int vec = se_pick_vec(C, (int)dx.element_size());
  const float invHW = 1.f / (float)HW;
#define LAUNCH(T, V)                                                            \
  se_bwd_add_pool_kernel<T, V><<<dfd_grid(NHW*(C / V), 256), 256, 0, stream>>>( \
      (T*)dx.data_ptr(), dsc.data_ptr<float>(), NHW, C, HW, invHW)
#define LV(T)                                                                   \
  switch (vec) {                                                                \
    case 8: LAUNCH(T, 8); break;                                                \
    case 4: LAUNCH(T, 4); break;                                                \
    case 2: LAUNCH(T, 2); break;                                                \
    default: LAUNCH(T, 1); break;                                               \
  }
  if (dx.scalar_type() == at::kBFloat16) { LV(__hip_bfloat16) }
  else if (dx.scalar_type() == at::kHalf) { LV(__half) }
  else if (dx.scalar_type() == at::kFloat) { LV(float) }
  else TORCH_CHECK(false, "se_bwd_add_pool: unsupported dtype");
#undef LV
#undef LAUNCH
}
