// Fused BatchNorm2d + activation (forward + backward), NHWC, gfx950.
//
// Replaces the reference's separate cuDNN BatchNorm + jit-scripted Swish
// kernels (reference dfd/timm/models/layers/activations.py:19-48; BN at
// every efficientnet block). Input/activation dtype
// bf16/fp16/fp32; all statistics and parameters fp32.
//
// Layout: channels_last (N,C,H,W) == row-major [M, C] with M = N*H*W and C
// contiguous. Every kernel uses the same block shape: a power-of-two number
// of channel-slots (cpb ≤ 64) × row-groups; each thread owns a VEC-wide
// contiguous channel slice, loads its per-channel fp32 coefficients into
// REGISTERS once, and streams rows. This keeps the hot loop at pure
// vectorized stream traffic — the first cut reloaded 6 fp32 coefficients
// per element from L1 and ran 8× off roofline
// (bn_act_bwd_dx 774 us vs se_bwd_reduce 90 us on the same bytes).
//
// Training forward is stats-reduce -> finalize -> fused normalize+act;
// backward is reduce (dgamma/dbeta with act' recompute) + elementwise dx.
// SiLU backward recomputes sigma(z) from saved mean/invstd (the "hard part"
// flagged in SURVEY.md §7).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T, int N>
struct alignas(sizeof(T) * N) BVec {
  T v[N];
};

template <typename T, int N>
DFD_DEV BVec<T, N> bvload(const T* p) {
  return *reinterpret_cast<const BVec<T, N>*>(p);
}

template <typename T, int N>
DFD_DEV void bvstore(T* p, const BVec<T, N>& x) {
  *reinterpret_cast<BVec<T, N>*>(p) = x;
}

// ---- block/grid plan shared by all kernels --------------------------------
int bn_pick_vec(long long c, int elem_size) {
  const int max_vec = elem_size == 4 ? 4 : 8;
  for (int v = max_vec; v > 1; v >>= 1)
    if (c % v == 0) return v;
  return 1;
}

struct BnPlan {
  int cpb, ctiles, chunks, rows_per_chunk;
};

// Channel-slot count per block is NOT rounded up to a power of two: C=144
// with VEC=8 (cv=18) previously ran with 18 of 32 slots active — 44% of
// every block's load bandwidth idle (r01->r02 profile, bn kernels ~3x off
// the streaming ceiling). Tiles are balanced across ctiles instead.
BnPlan bn_plan(int cv, long long rows) {
  BnPlan p;
  p.ctiles = (cv + 63) / 64;
  p.cpb = (cv + p.ctiles - 1) / p.ctiles;
  long long want = (2048 + p.ctiles - 1) / p.ctiles;
  const int nrg = 256 / p.cpb;
  // floor of ~16 row-iterations per thread: small tensors otherwise explode
  // into 2048 blocks whose per-block atomics serialize on C addresses
  long long by_iters = rows / ((long long)nrg * 16);
  if (want > by_iters) want = by_iters;
  long long max_chunks = (rows + nrg - 1) / nrg;
  if (want > max_chunks) want = max_chunks;
  if (want < 1) want = 1;
  p.chunks = (int)want;
  p.rows_per_chunk = (int)((rows + p.chunks - 1) / p.chunks);
  return p;
}

// ---------------------------------------------------------------------------
// stats reduce: sum and sumsq per channel (grid: ctiles × row-chunks)
// ---------------------------------------------------------------------------
template <typename T, int VEC>
__global__ void bn_stats_kernel(const T* __restrict__ x, float* __restrict__ sum,
                                float* __restrict__ sumsq, long long M, int C,
                                int cpb, int rows_per_chunk) {
  extern __shared__ float lds[];  // [256 * VEC]
  const int slot = threadIdx.x % cpb;
  const int rg = threadIdx.x / cpb;
  const int nrg = blockDim.x / cpb;
  const int cv = C / VEC;
  const int cvec = blockIdx.x * cpb + slot;
  // tail threads (cpb not dividing 256) must not stream rows: their rg
  // aliases row-group 0's stride pattern and would double-count
  const bool active = cvec < cv && rg < nrg;
  const int c = cvec * VEC;

  float s[VEC], q[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) { s[j] = 0.f; q[j] = 0.f; }

  if (active) {
    const long long r0 = (long long)blockIdx.y * rows_per_chunk;
    const long long r1 = min(r0 + rows_per_chunk, M);
    const T* xc = x + c;
    long long r = r0 + rg;
    // 4-row batches: independent loads issue together instead of one
    // load-wait-consume cycle per row
    for (; r + 3 * (long long)nrg < r1; r += 4 * (long long)nrg) {
      BVec<T, VEC> v[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) v[u] = bvload<T, VEC>(xc + (r + u * (long long)nrg) * C);
#pragma unroll
      for (int u = 0; u < 4; ++u)
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          const float f = DfdCvt<T>::to_f32(v[u].v[j]);
          s[j] += f;
          q[j] += f * f;
        }
    }
    for (; r < r1; r += nrg) {
      const BVec<T, VEC> xv = bvload<T, VEC>(xc + r * C);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        const float f = DfdCvt<T>::to_f32(xv.v[j]);
        s[j] += f;
        q[j] += f * f;
      }
    }
  }

  float* my = lds + (size_t)(rg * cpb + slot) * VEC;
  int p2 = 1;
  while (p2 * 2 <= nrg) p2 *= 2;
  const bool in_block = rg < nrg;  // cpb may not divide 256: tail threads idle
  // reduce s then q through the same LDS buffer
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    __syncthreads();
    if (in_block) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) my[j] = pass == 0 ? s[j] : q[j];
    }
    __syncthreads();
    if (in_block && rg >= p2) {  // fold the non-pow2 excess (unique dst)
      float* dst = lds + (size_t)((rg - p2) * cpb + slot) * VEC;
#pragma unroll
      for (int j = 0; j < VEC; ++j) dst[j] += my[j];
    }
    __syncthreads();
    for (int step = p2 >> 1; step > 0; step >>= 1) {
      if (rg < step) {
        const float* other = lds + ((size_t)((rg + step) * cpb) + slot) * VEC;
#pragma unroll
        for (int j = 0; j < VEC; ++j) my[j] += other[j];
      }
      __syncthreads();
    }
    if (rg == 0 && active) {
      float* out = pass == 0 ? sum : sumsq;
#pragma unroll
      for (int j = 0; j < VEC; ++j) atomicAdd(out + c + j, my[j]);
    }
  }
}

// ---------------------------------------------------------------------------
// bucket fold: sum the 64 bucketed partial buffers a producer kernel's stats
// epilogue wrote ([buckets, 2, C] -> sum[C], sumsq[C])
// ---------------------------------------------------------------------------
__global__ void bn_fold_buckets_kernel(const float* __restrict__ buckets,
                                       float* __restrict__ sum,
                                       float* __restrict__ sumsq, int C, int nb) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, q = 0.f;
  for (int b = 0; b < nb; ++b) {
    s += buckets[(size_t)b * 2 * C + c];
    q += buckets[(size_t)b * 2 * C + C + c];
  }
  sum[c] = s;
  sumsq[c] = q;
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
// scale/shift live in registers; rows stream.
// ---------------------------------------------------------------------------
template <typename T, Act ACT, int VEC, bool RES>
__global__ void bn_act_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                  const float* __restrict__ scale,
                                  const float* __restrict__ shift,
                                  const T* __restrict__ res,
                                  const float* __restrict__ dp,  // [B] per-sample
                                  long long hw,                  // rows per sample
                                  long long M, int C,
                                  int cpb, int rows_per_chunk) {
  const int slot = threadIdx.x % cpb;
  const int rg = threadIdx.x / cpb;
  const int nrg = blockDim.x / cpb;
  const int cv = C / VEC;
  const int cvec = blockIdx.x * cpb + slot;
  if (cvec >= cv || rg >= nrg) return;
  const int c = cvec * VEC;

  float sc[VEC], sh[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    sc[j] = scale[c + j];
    sh[j] = shift[c + j];
  }

  const long long r0 = (long long)blockIdx.y * rows_per_chunk;
  const long long r1 = min(r0 + rows_per_chunk, M);
  long long r = r0 + rg;
  // 8-row batches: up to 16 loads in flight per iteration (wait_any was
  // 52% at 2-row, ~50% at 4-row)
  for (; r + 7 * (long long)nrg < r1; r += 8 * (long long)nrg) {
    BVec<T, VEC> xv[8], rv[8], yv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      xv[u] = bvload<T, VEC>(x + (r + u * (long long)nrg) * C + c);
      if (RES) rv[u] = bvload<T, VEC>(res + (r + u * (long long)nrg) * C + c);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const float keep = dp ? dp[(r + u * (long long)nrg) / hw] : 1.f;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        const float z = fmaf(DfdCvt<T>::to_f32(xv[u].v[j]), sc[j], sh[j]);
        float v = act_fwd(z, ACT);
        if (dp) v *= keep;  // fused drop_path (stochastic depth) scale
        if (RES) v += DfdCvt<T>::to_f32(rv[u].v[j]);
        yv[u].v[j] = DfdCvt<T>::from_f32(v);
      }
      bvstore<T, VEC>(y + (r + u * (long long)nrg) * C + c, yv[u]);
    }
  }
  for (; r < r1; r += nrg) {
    const BVec<T, VEC> xv = bvload<T, VEC>(x + r * C + c);
    const float keep = dp ? dp[r / hw] : 1.f;
    BVec<T, VEC> yv;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      const float z = fmaf(DfdCvt<T>::to_f32(xv.v[j]), sc[j], sh[j]);
      float v = act_fwd(z, ACT);
      if (dp) v *= keep;
      if (RES) v += DfdCvt<T>::to_f32(res[r * C + c + j]);
      yv.v[j] = DfdCvt<T>::from_f32(v);
    }
    bvstore<T, VEC>(y + r * C + c, yv);
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
    const float* __restrict__ dp, long long hw, long long M, int C,
    int cpb, int rows_per_chunk) {
  extern __shared__ float lds[];
  const int slot = threadIdx.x % cpb;
  const int rg = threadIdx.x / cpb;
  const int nrg = blockDim.x / cpb;
  const int cv = C / VEC;
  const int cvec = blockIdx.x * cpb + slot;
  const bool active = cvec < cv && rg < nrg;
  const int c = cvec * VEC;

  float sg[VEC], sgx[VEC], mn[VEC], is[VEC], ga[VEC], be[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    sg[j] = 0.f;
    sgx[j] = 0.f;
    const int cc = active ? c + j : 0;
    mn[j] = mean[cc];
    is[j] = invstd[cc];
    ga[j] = weight ? weight[cc] : 1.f;
    be[j] = bias ? bias[cc] : 0.f;
  }

  if (active) {
    const long long r0 = (long long)blockIdx.y * rows_per_chunk;
    const long long r1 = min(r0 + rows_per_chunk, M);
    long long r = r0 + rg;
    // 4-row batches: 8 independent loads in flight per iteration (the 2-row
    // version measured 2.4x off the fwd kernel's streaming rate — r01 profile)
    for (; r + 3 * (long long)nrg < r1; r += 4 * (long long)nrg) {
      BVec<T, VEC> xv[4], dv[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        xv[u] = bvload<T, VEC>(x + (r + u * (long long)nrg) * C + c);
        dv[u] = bvload<T, VEC>(dy + (r + u * (long long)nrg) * C + c);
      }
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const float keep = dp ? dp[(r + u * (long long)nrg) / hw] : 1.f;
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          const float xh = (DfdCvt<T>::to_f32(xv[u].v[j]) - mn[j]) * is[j];
          const float z = fmaf(ga[j], xh, be[j]);
          float g = DfdCvt<T>::to_f32(dv[u].v[j]) * act_bwd(z, ACT);
          if (dp) g *= keep;
          sg[j] += g;
          sgx[j] += g * xh;
        }
      }
    }
    for (; r < r1; r += nrg) {
      const BVec<T, VEC> xv = bvload<T, VEC>(x + r * C + c);
      const BVec<T, VEC> dv = bvload<T, VEC>(dy + r * C + c);
      const float keep = dp ? dp[r / hw] : 1.f;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        const float xh = (DfdCvt<T>::to_f32(xv.v[j]) - mn[j]) * is[j];
        const float z = fmaf(ga[j], xh, be[j]);
        float g = DfdCvt<T>::to_f32(dv.v[j]) * act_bwd(z, ACT);
        if (dp) g *= keep;
        sg[j] += g;
        sgx[j] += g * xh;
      }
    }
  }

  float* my = lds + (size_t)(rg * cpb + slot) * VEC;
  int p2 = 1;
  while (p2 * 2 <= nrg) p2 *= 2;
  const bool in_block = rg < nrg;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    __syncthreads();
    if (in_block) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) my[j] = pass == 0 ? sg[j] : sgx[j];
    }
    __syncthreads();
    if (in_block && rg >= p2) {  // fold the non-pow2 excess (unique dst)
      float* dst = lds + (size_t)((rg - p2) * cpb + slot) * VEC;
#pragma unroll
      for (int j = 0; j < VEC; ++j) dst[j] += my[j];
    }
    __syncthreads();
    for (int step = p2 >> 1; step > 0; step >>= 1) {
      if (rg < step) {
        const float* other = lds + ((size_t)((rg + step) * cpb) + slot) * VEC;
#pragma unroll
        for (int j = 0; j < VEC; ++j) my[j] += other[j];
      }
      __syncthreads();
    }
    if (rg == 0 && active) {
      float* out = pass == 0 ? dbeta : dgamma;
#pragma unroll
      for (int j = 0; j < VEC; ++j) atomicAdd(out + c + j, my[j]);
    }
  }
}

// ---------------------------------------------------------------------------
// backward dx elementwise:
//   train: dx = k1*g - k2 - k3*(x - mean),  g = dy*act'(scale*x + shift)
//          k1 = ga*is, k2 = ga*is*dbeta/M, k3 = ga*is^2*dgamma/M
//   eval:  dx = k1*g
// ---------------------------------------------------------------------------
template <typename T, Act ACT, bool TRAIN, int VEC>
__global__ void bn_act_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ weight, const float* __restrict__ bias,
    const float* __restrict__ dgamma, const float* __restrict__ dbeta,
    const float* __restrict__ dp, long long hw,
    long long M, int C, float invM, int cpb, int rows_per_chunk) {
  const int slot = threadIdx.x % cpb;
  const int rg = threadIdx.x / cpb;
  const int nrg = blockDim.x / cpb;
  const int cv = C / VEC;
  const int cvec = blockIdx.x * cpb + slot;
  if (cvec >= cv || rg >= nrg) return;
  const int c = cvec * VEC;

  float sc[VEC], sh[VEC], mn[VEC], k1[VEC], k2[VEC], k3[VEC];
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    const int cc = c + j;
    const float is = invstd[cc];
    const float ga = weight ? weight[cc] : 1.f;
    mn[j] = mean[cc];
    sc[j] = ga * is;
    sh[j] = (bias ? bias[cc] : 0.f) - mn[j] * ga * is;
    k1[j] = ga * is;
    k2[j] = TRAIN ? ga * is * dbeta[cc] * invM : 0.f;
    k3[j] = TRAIN ? ga * is * is * dgamma[cc] * invM : 0.f;
  }

  const long long r0 = (long long)blockIdx.y * rows_per_chunk;
  const long long r1 = min(r0 + rows_per_chunk, M);
  long long r = r0 + rg;
  for (; r + nrg < r1; r += 2 * (long long)nrg) {
    BVec<T, VEC> xv[2], dv[2], ov[2];
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      xv[u] = bvload<T, VEC>(x + (r + u * (long long)nrg) * C + c);
      dv[u] = bvload<T, VEC>(dy + (r + u * (long long)nrg) * C + c);
    }
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      const float keep = dp ? dp[(r + u * (long long)nrg) / hw] : 1.f;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        const float xf = DfdCvt<T>::to_f32(xv[u].v[j]);
        const float z = fmaf(xf, sc[j], sh[j]);
        float g = DfdCvt<T>::to_f32(dv[u].v[j]) * act_bwd(z, ACT);
        if (dp) g *= keep;
        float v = k1[j] * g;
        if (TRAIN) v = v - k2[j] - k3[j] * (xf - mn[j]);
        ov[u].v[j] = DfdCvt<T>::from_f32(v);
      }
      bvstore<T, VEC>(dx + (r + u * (long long)nrg) * C + c, ov[u]);
    }
  }
  for (; r < r1; r += nrg) {
    const BVec<T, VEC> xv = bvload<T, VEC>(x + r * C + c);
    const BVec<T, VEC> dv = bvload<T, VEC>(dy + r * C + c);
    const float keep = dp ? dp[r / hw] : 1.f;
    BVec<T, VEC> ov;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      const float xf = DfdCvt<T>::to_f32(xv.v[j]);
      const float z = fmaf(xf, sc[j], sh[j]);
      float g = DfdCvt<T>::to_f32(dv.v[j]) * act_bwd(z, ACT);
      if (dp) g *= keep;
      float v = k1[j] * g;
      if (TRAIN) v = v - k2[j] - k3[j] * (xf - mn[j]);
      ov.v[j] = DfdCvt<T>::from_f32(v);
    }
    bvstore<T, VEC>(dx + r * C + c, ov);
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

#define DISPATCH_VEC(vec, ...)                                        \
  [&] {                                                               \
    if (vec == 8) {                                                   \
      constexpr int KVEC = 8;                                         \
      return __VA_ARGS__();                                           \
    } else if (vec == 4) {                                            \
      constexpr int KVEC = 4;                                         \
      return __VA_ARGS__();                                           \
    } else if (vec == 2) {                                            \
      constexpr int KVEC = 2;                                         \
      return __VA_ARGS__();                                           \
    } else {                                                          \
      constexpr int KVEC = 1;                                         \
      return __VA_ARGS__();                                           \
    }                                                                 \
  }()

}  // namespace

// x: (N,C,H,W) channels_last. Returns {y, save_mean, save_invstd}.
std::vector<at::Tensor> bn_act_fwd(
    at::Tensor x, at::Tensor weight, at::Tensor bias,
    at::Tensor running_mean, at::Tensor running_var,
    bool training, double momentum, double eps, std::string act_s,
    c10::optional<at::Tensor> residual_opt,
    c10::optional<at::Tensor> stats_opt,
    c10::optional<at::Tensor> drop_path_opt) {
  at::Tensor dp_t = drop_path_opt.has_value() ? *drop_path_opt : at::Tensor();
  const float* dp_p = nullptr;
  if (dp_t.defined()) {
    TORCH_CHECK(dp_t.scalar_type() == at::kFloat && dp_t.is_contiguous() &&
                    dp_t.numel() == x.size(0),
                "bn_act_fwd: drop_path mask must be fp32 [B]");
    dp_p = dp_t.data_ptr<float>();
  }
  at::Tensor residual = residual_opt.has_value() ? *residual_opt : at::Tensor();
  const bool has_res = residual.defined();
  if (has_res) {
    TORCH_CHECK(residual.sizes() == x.sizes() && residual.scalar_type() == x.scalar_type(),
                "bn_act_fwd: residual must match x");
    TORCH_CHECK(residual.is_contiguous(at::MemoryFormat::ChannelsLast),
                "bn_act_fwd: residual must be channels_last");
  }
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "bn_act_fwd: 4D CUDA tensor expected");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn_act_fwd: channels_last input required");
  const Act act = act_from_string(act_s);
  const int C = (int)x.size(1);
  const long long M = (long long)x.size(0) * x.size(2) * x.size(3);

  auto stream = at::cuda::getCurrentHIPStream();
  auto opts_f = x.options().dtype(at::kFloat);
  auto y = at::empty_like(x);
  auto save_mean = at::empty({C}, opts_f);
  auto save_invstd = at::empty({C}, opts_f);
  auto scale = at::empty({C}, opts_f);
  auto shift = at::empty({C}, opts_f);

  const float* w_p = weight.defined() ? weight.data_ptr<float>() : nullptr;
  const float* b_p = bias.defined() ? bias.data_ptr<float>() : nullptr;

  const int vec = bn_pick_vec(C, (int)x.element_size());
  const BnPlan plan = bn_plan(C / vec, M);
  dim3 grid(plan.ctiles, plan.chunks);
  const int lds = 256 * vec * sizeof(float);

  if (training) {
    auto sum = at::empty({C}, opts_f);
    auto sumsq = at::empty({C}, opts_f);
    if (stats_opt.has_value()) {
      // producer-fused path: the kernel that WROTE x already accumulated
      // per-channel sum/sumsq into 64 bucketed fp32 buffers — fold them and
      // skip the full extra read of x.
      at::Tensor buckets = *stats_opt;
      TORCH_CHECK(buckets.scalar_type() == at::kFloat && buckets.is_contiguous() &&
                      buckets.numel() % (2 * (long long)C) == 0,
                  "bn_act_fwd: stats buckets must be fp32 [nb, 2, C]");
      const int nb = (int)(buckets.numel() / (2 * (long long)C));
      hipLaunchKernelGGL(bn_fold_buckets_kernel, dim3((C + 255) / 256), dim3(256), 0,
                         stream, buckets.data_ptr<float>(), sum.data_ptr<float>(),
                         sumsq.data_ptr<float>(), C, nb);
    } else {
      sum.zero_();
      sumsq.zero_();
      DISPATCH_DTYPE(x.scalar_type(), "bn_stats", [&] {
        DISPATCH_VEC(vec, [&] {
          hipLaunchKernelGGL((bn_stats_kernel<T, KVEC>), grid, dim3(256), lds, stream,
                             (const T*)x.data_ptr(), sum.data_ptr<float>(),
                             sumsq.data_ptr<float>(), M, C, plan.cpb,
                             plan.rows_per_chunk);
        });
      });
    }
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

  DISPATCH_DTYPE(x.scalar_type(), "bn_act_fwd", [&] {
    DISPATCH_ACT(act, [&] {
      DISPATCH_VEC(vec, [&] {
        const long long hw = (long long)x.size(2) * x.size(3);
        if (has_res) {
          hipLaunchKernelGGL((bn_act_fwd_kernel<T, ACT, KVEC, true>), grid, dim3(256), 0,
                             stream, (const T*)x.data_ptr(), (T*)y.data_ptr(),
                             scale.data_ptr<float>(), shift.data_ptr<float>(),
                             (const T*)residual.data_ptr(), dp_p, hw, M, C,
                             plan.cpb, plan.rows_per_chunk);
        } else {
          hipLaunchKernelGGL((bn_act_fwd_kernel<T, ACT, KVEC, false>), grid, dim3(256), 0,
                             stream, (const T*)x.data_ptr(), (T*)y.data_ptr(),
                             scale.data_ptr<float>(), shift.data_ptr<float>(),
                             (const T*)nullptr, dp_p, hw, M, C,
                             plan.cpb, plan.rows_per_chunk);
        }
      });
    });
  });
  return {y, save_mean, save_invstd};
}

// Returns {dx, dgamma, dbeta}.
std::vector<at::Tensor> bn_act_bwd(
    at::Tensor dy, at::Tensor x, at::Tensor weight, at::Tensor bias,
    at::Tensor save_mean, at::Tensor save_invstd, bool training, std::string act_s,
    c10::optional<at::Tensor> drop_path_opt) {
  at::Tensor dp_t = drop_path_opt.has_value() ? *drop_path_opt : at::Tensor();
  const float* dp_p = nullptr;
  if (dp_t.defined()) {
    TORCH_CHECK(dp_t.scalar_type() == at::kFloat && dp_t.is_contiguous() &&
                    dp_t.numel() == x.size(0),
                "bn_act_bwd: drop_path mask must be fp32 [B]");
    dp_p = dp_t.data_ptr<float>();
  }
  const long long hw_bn = (long long)x.size(2) * x.size(3);
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4, "bn_act_bwd: 4D CUDA tensor expected");
  const Act act = act_from_string(act_s);
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int C = (int)x.size(1);
  const long long M = (long long)x.size(0) * x.size(2) * x.size(3);

  auto stream = at::cuda::getCurrentHIPStream();
  auto opts_f = x.options().dtype(at::kFloat);
  auto dgamma = at::zeros({C}, opts_f);
  auto dbeta = at::zeros({C}, opts_f);
  auto dx = at::empty_like(x);

  const float* w_p = weight.defined() ? weight.data_ptr<float>() : nullptr;
  const float* b_p = bias.defined() ? bias.data_ptr<float>() : nullptr;

  const int vec = bn_pick_vec(C, (int)x.element_size());
  const BnPlan plan = bn_plan(C / vec, M);
  dim3 grid(plan.ctiles, plan.chunks);
  const int lds = 256 * vec * sizeof(float);
  const float invM = 1.f / (float)M;

  DISPATCH_DTYPE(x.scalar_type(), "bn_bwd_reduce", [&] {
    DISPATCH_ACT(act, [&] {
      DISPATCH_VEC(vec, [&] {
        hipLaunchKernelGGL((bn_act_bwd_reduce_kernel<T, ACT, KVEC>), grid, dim3(256), lds,
                           stream, (const T*)dy.data_ptr(), (const T*)x.data_ptr(),
                           save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
                           w_p, b_p, dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                           dp_p, hw_bn, M, C, plan.cpb, plan.rows_per_chunk);
      });
    });
  });

  DISPATCH_DTYPE(x.scalar_type(), "bn_bwd_dx", [&] {
    DISPATCH_ACT(act, [&] {
      DISPATCH_VEC(vec, [&] {
        if (training) {
          hipLaunchKernelGGL((bn_act_bwd_dx_kernel<T, ACT, true, KVEC>), grid, dim3(256), 0,
                             stream, (const T*)dy.data_ptr(), (const T*)x.data_ptr(),
                             (T*)dx.data_ptr(), save_mean.data_ptr<float>(),
                             save_invstd.data_ptr<float>(), w_p, b_p,
                             dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                             dp_p, hw_bn, M, C, invM, plan.cpb, plan.rows_per_chunk);
        } else {
          hipLaunchKernelGGL((bn_act_bwd_dx_kernel<T, ACT, false, KVEC>), grid, dim3(256), 0,
                             stream, (const T*)dy.data_ptr(), (const T*)x.data_ptr(),
                             (T*)dx.data_ptr(), save_mean.data_ptr<float>(),
                             save_invstd.data_ptr<float>(), w_p, b_p,
                             dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                             dp_p, hw_bn, M, C, invM, plan.cpb, plan.rows_per_chunk);
        }
      });
    });
  });
  return {dx, dgamma, dbeta};
}
