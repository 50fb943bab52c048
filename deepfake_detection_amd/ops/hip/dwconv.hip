// Depthwise 2-D convolution for gfx950 (CDNA4), NHWC (torch channels_last).
//
// Replaces the MIOpen grouped-conv path for depthwise convs (groups == C):
// the reference delegates these to cuDNN (SURVEY.md §2.6 item 3,
// dfd/timm/models/layers/create_conv2d.py:24-25); on ROCm the MIOpen/CK
// grouped bwd-weight kernel dominated the whole training step (87% of GPU
// time, profiles/r01_bench_b4_299_bs192_top_kernels.md), so all three
// passes are hand-written here.
//
// Layouts:
//   x, y, dy, dx : NHWC, C fastest (channels_last), dtype bf16/fp16/fp32
//   weight       : repacked by the python wrapper to (KH, KW, C), same dtype
//   dweight      : (KH, KW, C) fp32 (wrapper casts/permutes back)
//
// Design: bandwidth-bound op. Each thread owns a VEC-wide contiguous channel
// slice (VEC chosen so C % VEC == 0; 16 B loads for bf16 at VEC=8) and
// accumulates in fp32 registers. bwd-weight blocks are (kh, channel-vector)
// grids of threads looping over a spatial chunk, finishing with one fp32
// atomicAdd per (kh,kw,c) — K*VEC atomics per thread total.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T, int N>
struct alignas(sizeof(T) * N) TVec {
  T v[N];
};

template <typename T, int N>
DFD_DEV TVec<T, N> vload(const T* p) {
  return *reinterpret_cast<const TVec<T, N>*>(p);
}

template <typename T, int N>
DFD_DEV void vstore(T* p, const TVec<T, N>& x) {
  *reinterpret_cast<TVec<T, N>*>(p) = x;
}

// ---------------------------------------------------------------------------
// forward: y[n,ho,wo,c] = sum_kh,kw x[n, ho*sh-ph+kh, wo*sw-pw+kw, c] * w[kh,kw,c]
// ---------------------------------------------------------------------------
template <typename T, int K, int VEC>
__global__ void dw_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                              T* __restrict__ y, int N, int C, int H, int W,
                              int Ho, int Wo, int sh, int sw, int ph, int pw) {
  const int cv = C / VEC;
  const long long total = (long long)N * Ho * Wo * cv;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(idx % cv) * VEC;
    long long p = idx / cv;
    const int wo = (int)(p % Wo);
    p /= Wo;
    const int ho = (int)(p % Ho);
    const int n = (int)(p / Ho);

    float acc[VEC];
#pragma unroll
    for (int i = 0; i < VEC; ++i) acc[i] = 0.f;

    const int hi0 = ho * sh - ph;
    const int wi0 = wo * sw - pw;
    const bool interior = wi0 >= 0 && wi0 + K - 1 < W;
#pragma unroll
    for (int kh = 0; kh < K; ++kh) {
      const int hi = hi0 + kh;
      if (hi < 0 || hi >= H) continue;
      const T* xrow = x + (((long long)n * H + hi) * W) * C + c;
      if (interior) {
        TVec<T, VEC> xv[K];
#pragma unroll
        for (int kw = 0; kw < K; ++kw)
          xv[kw] = vload<T, VEC>(xrow + (long long)(wi0 + kw) * C);
#pragma unroll
        for (int kw = 0; kw < K; ++kw) {
          const TVec<T, VEC> wv = vload<T, VEC>(w + ((long long)kh * K + kw) * C + c);
#pragma unroll
          for (int i = 0; i < VEC; ++i)
            acc[i] += DfdCvt<T>::to_f32(xv[kw].v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
        }
      } else {
#pragma unroll
        for (int kw = 0; kw < K; ++kw) {
          const int wi = wi0 + kw;
          if (wi < 0 || wi >= W) continue;
          const TVec<T, VEC> xv = vload<T, VEC>(xrow + (long long)wi * C);
          const TVec<T, VEC> wv = vload<T, VEC>(w + ((long long)kh * K + kw) * C + c);
#pragma unroll
          for (int i = 0; i < VEC; ++i)
            acc[i] += DfdCvt<T>::to_f32(xv.v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
        }
      }
    }
    TVec<T, VEC> yv;
#pragma unroll
    for (int i = 0; i < VEC; ++i) yv.v[i] = DfdCvt<T>::from_f32(acc[i]);
    vstore<T, VEC>(y + (((long long)n * Ho + ho) * Wo + wo) * C + c, yv);
  }
}

// ---------------------------------------------------------------------------
// NOTE (r02): this body is the round-1 kernel RESTORED VERBATIM. Five
// re-tilings were measured against it on MI355X (2-row tiles, wider TW,
// software-pipelined row loads, register-resident weights, per-K hybrids)
// and every one lost in the full bench — the kernel is latency-bound and
// extremely sensitive to compiler scheduling/occupancy; see BASELINE.md
// "next levers" for the remaining ideas (LDS-staged weights).
// forward, stride 1, TW consecutive outputs per thread.
// Loads per output drop from K*K to ~K*(K+TW-1)/TW (x) and weight loads
// amortize by TW; bwd-data for stride 1 reuses this kernel with the packed
// weight flipped in (kh,kw) and padding (K-1-p) — it is the same stride-1
// correlation.
// ---------------------------------------------------------------------------
template <typename T, int K, int VEC, int TW>
__global__ void dw_fwd_s1_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                 T* __restrict__ y, int N, int C, int H, int W,
                                 int Ho, int Wo, int ph, int pw) {
  const int cv = C / VEC;
  const int wt = (Wo + TW - 1) / TW;  // wo tiles per row
  const long long total = (long long)N * Ho * wt * cv;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(idx % cv) * VEC;
    long long p = idx / cv;
    const int wo0 = (int)(p % wt) * TW;
    p /= wt;
    const int ho = (int)(p % Ho);
    const int n = (int)(p / Ho);

    float acc[TW][VEC];
#pragma unroll
    for (int t = 0; t < TW; ++t)
#pragma unroll
      for (int i = 0; i < VEC; ++i) acc[t][i] = 0.f;

    const int hi0 = ho - ph;
    const int wi0 = wo0 - pw;
    const bool interior = wi0 >= 0 && wi0 + K + TW - 2 < W;
#pragma unroll
    for (int kh = 0; kh < K; ++kh) {
      const int hi = hi0 + kh;
      if (hi < 0 || hi >= H) continue;
      const T* xrow = x + (((long long)n * H + hi) * W) * C + c;
      const T* wrow = w + ((long long)kh * K) * C + c;
      if (interior) {
        // branchless: issue all K+TW-1 column loads, then consume (per-load
        // guards would serialize each load behind s_waitcnt vmcnt(0))
        TVec<T, VEC> xv[K + TW - 1];
#pragma unroll
        for (int col = 0; col < K + TW - 1; ++col)
          xv[col] = vload<T, VEC>(xrow + (long long)(wi0 + col) * C);
#pragma unroll
        for (int col = 0; col < K + TW - 1; ++col) {
#pragma unroll
          for (int t = 0; t < TW; ++t) {
            const int kw = col - t;
            if (kw < 0 || kw >= K) continue;
            const TVec<T, VEC> wv = vload<T, VEC>(wrow + (long long)kw * C);
#pragma unroll
            for (int i = 0; i < VEC; ++i)
              acc[t][i] += DfdCvt<T>::to_f32(xv[col].v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
          }
        }
      } else {
#pragma unroll
        for (int col = 0; col < K + TW - 1; ++col) {
          const int wi = wi0 + col;
          if (wi < 0 || wi >= W) continue;
          const TVec<T, VEC> xv = vload<T, VEC>(xrow + (long long)wi * C);
          // this column contributes to outputs t with 0 <= col - t < K
#pragma unroll
          for (int t = 0; t < TW; ++t) {
            const int kw = col - t;
            if (kw < 0 || kw >= K) continue;
            const TVec<T, VEC> wv = vload<T, VEC>(wrow + (long long)kw * C);
#pragma unroll
            for (int i = 0; i < VEC; ++i)
              acc[t][i] += DfdCvt<T>::to_f32(xv.v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
          }
        }
      }
    }
    T* yrow = y + (((long long)n * Ho + ho) * Wo) * C + c;
#pragma unroll
    for (int t = 0; t < TW; ++t) {
      if (wo0 + t >= Wo) break;
      TVec<T, VEC> yv;
#pragma unroll
      for (int i = 0; i < VEC; ++i) yv.v[i] = DfdCvt<T>::from_f32(acc[t][i]);
      vstore<T, VEC>(yrow + (long long)(wo0 + t) * C, yv);
    }
  }
}

// ---------------------------------------------------------------------------
// stats-emitting stride-1 forward (k3): bn-style (channel-slot, row-group)
// block layout — each thread's channel slice is FIXED, so the per-channel
// sum/sumsq of y fold in LDS and land in the bucketed stats buffers the
// following BatchNorm consumes (skipping its stats pass over y). k5 is
// excluded: a fixed channel makes the compiler hoist the whole K*K weight
// tile (k5: 100 VGPRs) and the occupancy loss measured slower than the
// separate stats pass it saves.
// ---------------------------------------------------------------------------
constexpr int kDwStatsBuckets = 64;

template <typename T, int K, int VEC, int TW>
__global__ void dw_fwd_s1_stats_kernel(
    const T* __restrict__ x, const T* __restrict__ w, T* __restrict__ y,
    float* __restrict__ stats,  // [kDwStatsBuckets, 2, C]
    int N, int C, int H, int W, int Ho, int Wo, int ph, int pw,
    int cpb, long long rows_per_chunk) {
  extern __shared__ float lds[];  // [256 * VEC] fp32 + K*K*cpb*VEC bf16 w slice
  T* wl = reinterpret_cast<T*>(lds + 256 * VEC);
  const int slot = threadIdx.x % cpb;
  const int rg = threadIdx.x / cpb;
  const int nrg = blockDim.x / cpb;
  const int cv = C / VEC;
  const int cvec = blockIdx.x * cpb + slot;
  const bool active = cvec < cv && rg < nrg;
  const int c = cvec * VEC;
  const int wt = (Wo + TW - 1) / TW;
  const long long rows = (long long)N * Ho * wt;

  // stage THIS block's channel slice of the weights: fixed-c threads would
  // otherwise make the compiler hoist the K*K tile into registers (the
  // occupancy loss measured 2x slower, r02); dynamic LDS reads stay cheap.
  const int c0 = blockIdx.x * cpb * VEC;
  const int cw = min(cpb * VEC, C - c0);  // slice width (elements)
  for (int i = threadIdx.x; i < K * K * cw; i += blockDim.x) {
    const int tap = i / cw;
    const int cc = i - tap * cw;
    wl[tap * (cpb * VEC) + cc] = w[(long long)tap * C + c0 + cc];
  }
  __syncthreads();

  float s[VEC], q[VEC];
#pragma unroll
  for (int i = 0; i < VEC; ++i) { s[i] = 0.f; q[i] = 0.f; }

  if (active) {
    const long long r0 = (long long)blockIdx.y * rows_per_chunk;
    const long long r1 = min(r0 + rows_per_chunk, rows);
    for (long long r = r0 + rg; r < r1; r += nrg) {
      const int wo0 = (int)(r % wt) * TW;
      long long p = r / wt;
      const int ho = (int)(p % Ho);
      const int n = (int)(p / Ho);

      float acc[TW][VEC];
#pragma unroll
      for (int t = 0; t < TW; ++t)
#pragma unroll
        for (int i = 0; i < VEC; ++i) acc[t][i] = 0.f;

      const int hi0 = ho - ph;
      const int wi0 = wo0 - pw;
      const bool interior = wi0 >= 0 && wi0 + K + TW - 2 < W;
#pragma unroll
      for (int kh = 0; kh < K; ++kh) {
        const int hi = hi0 + kh;
        if (hi < 0 || hi >= H) continue;
        const T* xrow = x + (((long long)n * H + hi) * W) * C + c;
        const T* wrow = wl + (kh * K) * (cpb * VEC) + slot * VEC;
        if (interior) {
          TVec<T, VEC> xv[K + TW - 1];
#pragma unroll
          for (int col = 0; col < K + TW - 1; ++col)
            xv[col] = vload<T, VEC>(xrow + (long long)(wi0 + col) * C);
#pragma unroll
          for (int col = 0; col < K + TW - 1; ++col)
#pragma unroll
            for (int t = 0; t < TW; ++t) {
              const int kw = col - t;
              if (kw < 0 || kw >= K) continue;
              const TVec<T, VEC> wv = vload<T, VEC>(wrow + kw * (cpb * VEC));
#pragma unroll
              for (int i = 0; i < VEC; ++i)
                acc[t][i] += DfdCvt<T>::to_f32(xv[col].v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
            }
        } else {
#pragma unroll
          for (int col = 0; col < K + TW - 1; ++col) {
            const int wi = wi0 + col;
            if (wi < 0 || wi >= W) continue;
            const TVec<T, VEC> xv = vload<T, VEC>(xrow + (long long)wi * C);
#pragma unroll
            for (int t = 0; t < TW; ++t) {
              const int kw = col - t;
              if (kw < 0 || kw >= K) continue;
              const TVec<T, VEC> wv = vload<T, VEC>(wrow + kw * (cpb * VEC));
#pragma unroll
              for (int i = 0; i < VEC; ++i)
                acc[t][i] += DfdCvt<T>::to_f32(xv.v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
            }
          }
        }
      }
      T* yrow = y + (((long long)n * Ho + ho) * Wo) * C + c;
#pragma unroll
      for (int t = 0; t < TW; ++t) {
        if (wo0 + t >= Wo) break;
        TVec<T, VEC> yv;
#pragma unroll
        for (int i = 0; i < VEC; ++i) {
          yv.v[i] = DfdCvt<T>::from_f32(acc[t][i]);
          const float f = DfdCvt<T>::to_f32(yv.v[i]);  // rounded value
          s[i] += f;
          q[i] += f * f;
        }
        vstore<T, VEC>(yrow + (long long)(wo0 + t) * C, yv);
      }
    }
  }

  // fold row-groups in LDS (non-pow2-aware), one bucketed atomic set per block
  float* my = lds + (size_t)(rg * cpb + slot) * VEC;
  int p2 = 1;
  while (p2 * 2 <= nrg) p2 *= 2;
  const bool in_block = rg < nrg;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    __syncthreads();
    if (in_block) {
#pragma unroll
      for (int i = 0; i < VEC; ++i) my[i] = pass == 0 ? s[i] : q[i];
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
      float* out = stats + (size_t)(blockIdx.y & (kDwStatsBuckets - 1)) * 2 * C +
                   (pass == 0 ? 0 : C);
#pragma unroll
      for (int i = 0; i < VEC; ++i) atomicAdd(out + c + i, my[i]);
    }
  }
}

// ---------------------------------------------------------------------------
// stride-1 forward with LDS-STAGED weights (BASELINE "next levers" item):
// the whole (K, K, C) packed weight is copied into LDS once per block and
// read with DYNAMIC offsets — the compiler can neither hoist it into
// registers (the k5 spill trap) nor send the per-tap reads through the
// vmem queue, which the x loads saturate (74% SQ_WAIT_ANY). Used when the
// weight fits a fraction of LDS that keeps >=3 blocks/CU resident.
// ---------------------------------------------------------------------------
template <typename T, int K, int VEC, int TW>
__global__ void dw_fwd_s1_ldsw_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                      T* __restrict__ y, int N, int C, int H, int W,
                                      int Ho, int Wo, int ph, int pw) {
  extern __shared__ unsigned char smem_raw[];
  T* wl = reinterpret_cast<T*>(smem_raw);  // [K*K*C]
  const int wtot = K * K * C;
  for (int i = threadIdx.x * VEC; i < wtot; i += blockDim.x * VEC) {
    if (i + VEC <= wtot)
      *reinterpret_cast<TVec<T, VEC>*>(&wl[i]) = vload<T, VEC>(w + i);
    else
      for (int e = i; e < wtot; ++e) wl[e] = w[e];
  }
  __syncthreads();

  const int cv = C / VEC;
  const int wt = (Wo + TW - 1) / TW;
  const long long total = (long long)N * Ho * wt * cv;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(idx % cv) * VEC;
    long long p = idx / cv;
    const int wo0 = (int)(p % wt) * TW;
    p /= wt;
    const int ho = (int)(p % Ho);
    const int n = (int)(p / Ho);

    float acc[TW][VEC];
#pragma unroll
    for (int t = 0; t < TW; ++t)
#pragma unroll
      for (int i = 0; i < VEC; ++i) acc[t][i] = 0.f;

    const int hi0 = ho - ph;
    const int wi0 = wo0 - pw;
    const bool interior = wi0 >= 0 && wi0 + K + TW - 2 < W;
#pragma unroll
    for (int kh = 0; kh < K; ++kh) {
      const int hi = hi0 + kh;
      if (hi < 0 || hi >= H) continue;
      const T* xrow = x + (((long long)n * H + hi) * W) * C + c;
      const T* wrow = wl + (kh * K) * C + c;
      if (interior) {
        TVec<T, VEC> xv[K + TW - 1];
#pragma unroll
        for (int col = 0; col < K + TW - 1; ++col)
          xv[col] = vload<T, VEC>(xrow + (long long)(wi0 + col) * C);
#pragma unroll
        for (int col = 0; col < K + TW - 1; ++col)
#pragma unroll
          for (int t = 0; t < TW; ++t) {
            const int kw = col - t;
            if (kw < 0 || kw >= K) continue;
            const TVec<T, VEC> wv = vload<T, VEC>(wrow + kw * C);
#pragma unroll
            for (int i = 0; i < VEC; ++i)
              acc[t][i] += DfdCvt<T>::to_f32(xv[col].v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
          }
      } else {
#pragma unroll
        for (int col = 0; col < K + TW - 1; ++col) {
          const int wi = wi0 + col;
          if (wi < 0 || wi >= W) continue;
          const TVec<T, VEC> xv = vload<T, VEC>(xrow + (long long)wi * C);
#pragma unroll
          for (int t = 0; t < TW; ++t) {
            const int kw = col - t;
            if (kw < 0 || kw >= K) continue;
            const TVec<T, VEC> wv = vload<T, VEC>(wrow + kw * C);
#pragma unroll
            for (int i = 0; i < VEC; ++i)
              acc[t][i] += DfdCvt<T>::to_f32(xv.v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
          }
        }
      }
    }
    T* yrow = y + (((long long)n * Ho + ho) * Wo) * C + c;
#pragma unroll
    for (int t = 0; t < TW; ++t) {
      if (wo0 + t >= Wo) break;
      TVec<T, VEC> yv;
#pragma unroll
      for (int i = 0; i < VEC; ++i) yv.v[i] = DfdCvt<T>::from_f32(acc[t][i]);
      vstore<T, VEC>(yrow + (long long)(wo0 + t) * C, yv);
    }
  }
}

// ---------------------------------------------------------------------------
// backward data:
// dx[n,hi,wi,c] = sum over (kh,kw) with hi = ho*sh-ph+kh solvable:
//                 dy[n,ho,wo,c] * w[kh,kw,c]
// ---------------------------------------------------------------------------
template <typename T, int K, int VEC>
__global__ void dw_bwd_data_kernel(const T* __restrict__ dy, const T* __restrict__ w,
                                   T* __restrict__ dx, int N, int C, int H, int W,
                                   int Ho, int Wo, int sh, int sw, int ph, int pw) {
  const int cv = C / VEC;
  const long long total = (long long)N * H * W * cv;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long long)gridDim.x * blockDim.x) {
    const int c = (int)(idx % cv) * VEC;
    long long p = idx / cv;
    const int wi = (int)(p % W);
    p /= W;
    const int hi = (int)(p % H);
    const int n = (int)(p / H);

    float acc[VEC];
#pragma unroll
    for (int i = 0; i < VEC; ++i) acc[i] = 0.f;

#pragma unroll
    for (int kh = 0; kh < K; ++kh) {
      const int th = hi + ph - kh;
      if (th < 0 || th % sh) continue;
      const int ho = th / sh;
      if (ho >= Ho) continue;
#pragma unroll
      for (int kw = 0; kw < K; ++kw) {
        const int tw = wi + pw - kw;
        if (tw < 0 || tw % sw) continue;
        const int wo = tw / sw;
        if (wo >= Wo) continue;
        const TVec<T, VEC> gv = vload<T, VEC>(dy + (((long long)n * Ho + ho) * Wo + wo) * C + c);
        const TVec<T, VEC> wv = vload<T, VEC>(w + ((long long)kh * K + kw) * C + c);
#pragma unroll
        for (int i = 0; i < VEC; ++i)
          acc[i] += DfdCvt<T>::to_f32(gv.v[i]) * DfdCvt<T>::to_f32(wv.v[i]);
      }
    }
    TVec<T, VEC> ov;
#pragma unroll
    for (int i = 0; i < VEC; ++i) ov.v[i] = DfdCvt<T>::from_f32(acc[i]);
    vstore<T, VEC>(dx + (((long long)n * H + hi) * W + wi) * C + c, ov);
  }
}

// ---------------------------------------------------------------------------
// backward weight:
// dw[kh,kw,c] = sum_n,ho,wo dy[n,ho,wo,c] * x[n, ho*sh-ph+kh, wo*sw-pw+kw, c]
//
// Block layout: 256 threads split as (channel-slot, kh, row-group):
//   slot = tid & (cpb-1)            cpb = power-of-two slots sized from C/VEC
//   rest = tid >> log2_cpb;  kh = rest % K;  row-group = rest / K
// Consecutive threads touch consecutive channel vectors (coalesced); the
// slot count adapts to C so small-C layers (C=48 → 6 vectors) still fill
// the block with row parallelism instead of idling 58/64 lanes (the
// fixed-64-slot first cut ran 1.15 ms on the k3 layers).
// Each thread owns one kernel row kh, loops its rows × all wo keeping K*VEC
// fp32 partials in registers, then one atomicAdd per partial.
// ---------------------------------------------------------------------------
template <typename T, int K, int VEC>
__global__ void dw_bwd_weight_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                     float* __restrict__ dw, int N, int C, int H, int W,
                                     int Ho, int Wo, int sh, int sw, int ph, int pw,
                                     int cpb, int rows_per_chunk) {
  extern __shared__ float lds[];  // [blockDim.x * VEC]
  const int slot = threadIdx.x % cpb;
  const int rest = threadIdx.x / cpb;
  const int kh = rest % K;
  const int rg = rest / K;
  const int nrg = (blockDim.x / cpb) / K;
  const int cv = C / VEC;
  const int cvec = blockIdx.x * cpb + slot;
  // no early return: every thread reaches the barriers below
  const bool active = (rg < nrg) && (cvec < cv);
  const int c = cvec * VEC;

  float acc[K * VEC];
#pragma unroll
  for (int i = 0; i < K * VEC; ++i) acc[i] = 0.f;

  if (active) {
    const long long rows_total = (long long)N * Ho;
    const long long row0 = (long long)blockIdx.y * rows_per_chunk;
    const long long row1 = min(row0 + rows_per_chunk, rows_total);
    // interior range where all K columns are in-bounds: guarded per-kw
    // branches otherwise force the compiler to serialize every load behind
    // s_waitcnt vmcnt(0) (measured 10x off the VALU bound)
    const int wo_lo = min(Wo, max(0, (pw + sw - 1) / sw));
    const int wo_hi = max(wo_lo, min(Wo, (W - K + pw) / sw + 1));

    for (long long r = row0 + rg; r < row1; r += nrg) {
      const int ho = (int)(r % Ho);
      const int n = (int)(r / Ho);
      const int hi = ho * sh - ph + kh;
      if (hi < 0 || hi >= H) continue;
      const T* dy_row = dy + (((long long)n * Ho + ho) * Wo) * C + c;
      const T* x_row = x + (((long long)n * H + hi) * W) * C + c;

      auto guarded = [&](int wo) {
        const TVec<T, VEC> gv = vload<T, VEC>(dy_row + (long long)wo * C);
        float gf[VEC];
#pragma unroll
        for (int i = 0; i < VEC; ++i) gf[i] = DfdCvt<T>::to_f32(gv.v[i]);
        const int wi0 = wo * sw - pw;
#pragma unroll
        for (int kw = 0; kw < K; ++kw) {
          const int wi = wi0 + kw;
          if (wi < 0 || wi >= W) continue;
          const TVec<T, VEC> xv = vload<T, VEC>(x_row + (long long)wi * C);
#pragma unroll
          for (int i = 0; i < VEC; ++i)
            acc[kw * VEC + i] += gf[i] * DfdCvt<T>::to_f32(xv.v[i]);
        }
      };
      for (int wo = 0; wo < wo_lo; ++wo) guarded(wo);
      for (int wo = wo_lo; wo < wo_hi; ++wo) {
        // branchless interior: issue dy + all K x loads, then consume
        const TVec<T, VEC> gv = vload<T, VEC>(dy_row + (long long)wo * C);
        const T* xp = x_row + (long long)(wo * sw - pw) * C;
        TVec<T, VEC> xv[K];
#pragma unroll
        for (int kw = 0; kw < K; ++kw) xv[kw] = vload<T, VEC>(xp + (long long)kw * C);
        float gf[VEC];
#pragma unroll
        for (int i = 0; i < VEC; ++i) gf[i] = DfdCvt<T>::to_f32(gv.v[i]);
#pragma unroll
        for (int kw = 0; kw < K; ++kw)
#pragma unroll
          for (int i = 0; i < VEC; ++i)
            acc[kw * VEC + i] += gf[i] * DfdCvt<T>::to_f32(xv[kw].v[i]);
      }
      for (int wo = wo_hi; wo < Wo; ++wo) guarded(wo);
    }
  }

  // fold the nrg row-groups in LDS (one kw plane at a time), then STORE the
  // block's partial to dw_part[chunk] — no atomics. A fold-then-atomic cut
  // still issued K*K*C atomics × 2048 blocks (≈13M) and fp32 global-atomic
  // throughput capped the kernel (~1.2 ms on C=288 k5); partial stores +
  // a tiny second-stage reduction run at stream rate.
  float* my = lds + (size_t)((kh + K * rg) * cpb + slot) * VEC;
  const int pow2 = 1 << (31 - __clz(nrg > 0 ? nrg : 1));
  float* part = dw + (long long)blockIdx.y * K * K * C;  // dw is [chunks,K,K,C]
  for (int kw = 0; kw < K; ++kw) {
    __syncthreads();
    if (rg < nrg) {
#pragma unroll
      for (int i = 0; i < VEC; ++i) my[i] = acc[kw * VEC + i];
    }
    __syncthreads();
    for (int step = pow2; step > 0; step >>= 1) {
      if (rg < step && rg + step < nrg) {
        const float* other =
            lds + ((size_t)((kh + K * (rg + step)) * cpb + slot)) * VEC;
#pragma unroll
        for (int i = 0; i < VEC; ++i) my[i] += other[i];
      }
      __syncthreads();
    }
    if (rg == 0 && active) {
#pragma unroll
      for (int i = 0; i < VEC; ++i)
        part[((long long)kh * K + kw) * C + c + i] = my[i];
    }
  }
}

// stage 2: dw[i] = sum_z part[z][i], i in [0, K*K*C). z is split across
// blockIdx.y (K*K*C alone is only ~7k threads — not enough to fill the chip
// when chunks is large), partial sums combine with a few atomics per element.
__global__ void dw_bwd_weight_reduce_kernel(const float* __restrict__ part,
                                            float* __restrict__ dw, long long kkc,
                                            int chunks, int z_per_blk) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= kkc) return;
  const int z0 = blockIdx.y * z_per_blk;
  const int z1 = min(z0 + z_per_blk, chunks);
  float s = 0.f;
  for (int z = z0; z < z1; ++z) s += part[(long long)z * kkc + i];
  if (gridDim.y == 1)
    dw[i] = s;
  else
    atomicAdd(dw + i, s);
}

// ---------------------------------------------------------------------------
// host-side dispatch
// ---------------------------------------------------------------------------

int pick_vec(long long c, int elem_size) {
  const int max_vec = elem_size == 4 ? 4 : 8;  // 16 B per lane load
  for (int v = max_vec; v > 1; v >>= 1)
    if (c % v == 0) return v;
  return 1;
}

struct Geom {
  int N, C, H, W, Ho, Wo, sh, sw, ph, pw;
};

template <typename T, int K, int VEC>
void launch_fwd(const at::Tensor& x, const at::Tensor& w, at::Tensor& y, const Geom& g,
                hipStream_t stream) {
  const int block = 256;
  if (g.sh == 1 && g.sw == 1) {
    constexpr int TW = 4;
    const int wt = (g.Wo + TW - 1) / TW;
    const long long total = (long long)g.N * g.Ho * wt * (g.C / VEC);
    const int wbytes = K * K * g.C * (int)x.element_size();
    if (wbytes <= 50 * 1024) {  // >=3 blocks/CU with the staged weight
      dw_fwd_s1_ldsw_kernel<T, K, VEC, TW>
          <<<dfd_grid(total, block), block, wbytes, stream>>>(
              (const T*)x.data_ptr(), (const T*)w.data_ptr(), (T*)y.data_ptr(),
              g.N, g.C, g.H, g.W, g.Ho, g.Wo, g.ph, g.pw);
      return;
    }
    dw_fwd_s1_kernel<T, K, VEC, TW><<<dfd_grid(total, block), block, 0, stream>>>(
        (const T*)x.data_ptr(), (const T*)w.data_ptr(), (T*)y.data_ptr(), g.N, g.C, g.H,
        g.W, g.Ho, g.Wo, g.ph, g.pw);
    return;
  }
  const long long total = (long long)g.N * g.Ho * g.Wo * (g.C / VEC);
  dw_fwd_kernel<T, K, VEC><<<dfd_grid(total, block), block, 0, stream>>>(
      (const T*)x.data_ptr(), (const T*)w.data_ptr(), (T*)y.data_ptr(), g.N, g.C, g.H,
      g.W, g.Ho, g.Wo, g.sh, g.sw, g.ph, g.pw);
}

template <typename T>
void fwd_ktype(const at::Tensor& x, const at::Tensor& w, at::Tensor& y, const Geom& g,
               int K, int vec, hipStream_t stream) {
  switch (K) {
#define DFD_CASE_K(KK)                                                      \
  case KK:                                                                  \
    switch (vec) {                                                          \
      case 8: launch_fwd<T, KK, 8>(x, w, y, g, stream); break;              \
      case 4: launch_fwd<T, KK, 4>(x, w, y, g, stream); break;              \
      case 2: launch_fwd<T, KK, 2>(x, w, y, g, stream); break;              \
      default: launch_fwd<T, KK, 1>(x, w, y, g, stream); break;             \
    }                                                                       \
    break;
    DFD_CASE_K(3)
    DFD_CASE_K(5)
    DFD_CASE_K(7)
    DFD_CASE_K(9)
    DFD_CASE_K(11)
#undef DFD_CASE_K
    default:
      TORCH_CHECK(false, "dwconv fwd: unsupported kernel size ", K);
  }
}

template <typename T, int K, int VEC>
void launch_bwd_data(const at::Tensor& dy, const at::Tensor& w, at::Tensor& dx,
                     const Geom& g, hipStream_t stream) {
  const long long total = (long long)g.N * g.H * g.W * (g.C / VEC);
  const int block = 256;
  dw_bwd_data_kernel<T, K, VEC><<<dfd_grid(total, block), block, 0, stream>>>(
      (const T*)dy.data_ptr(), (const T*)w.data_ptr(), (T*)dx.data_ptr(), g.N, g.C, g.H,
      g.W, g.Ho, g.Wo, g.sh, g.sw, g.ph, g.pw);
}

template <typename T>
void bwd_data_ktype(const at::Tensor& dy, const at::Tensor& w, at::Tensor& dx,
                    const Geom& g, int K, int vec, hipStream_t stream) {
  switch (K) {
#define DFD_CASE_K(KK)                                                      \
  case KK:                                                                  \
    switch (vec) {                                                          \
      case 8: launch_bwd_data<T, KK, 8>(dy, w, dx, g, stream); break;       \
      case 4: launch_bwd_data<T, KK, 4>(dy, w, dx, g, stream); break;       \
      case 2: launch_bwd_data<T, KK, 2>(dy, w, dx, g, stream); break;       \
      default: launch_bwd_data<T, KK, 1>(dy, w, dx, g, stream); break;      \
    }                                                                       \
    break;
    DFD_CASE_K(3)
    DFD_CASE_K(5)
    DFD_CASE_K(7)
    DFD_CASE_K(9)
    DFD_CASE_K(11)
#undef DFD_CASE_K
    default:
      TORCH_CHECK(false, "dwconv bwd_data: unsupported kernel size ", K);
  }
}

template <typename T, int K, int VEC>
void launch_bwd_weight(const at::Tensor& dy, const at::Tensor& x, at::Tensor& dw,
                       const Geom& g, hipStream_t stream) {
  const int cv = g.C / VEC;
  // exact (non-pow2) balanced channel slots per block, capped so at least
  // one (kh, row-group) pair fits (cpb * K <= 256) — the pow2 rounding both
  // idled slots AND forced extra channel tiles, each of which re-reads the
  // WHOLE dy+x (C=336 k5: 2 tiles of 32 instead of 1 of 42)
  const int cpb_cap = std::min(64, 256 / K);
  const int grid_x = (cv + cpb_cap - 1) / cpb_cap;
  const int cpb = (cv + grid_x - 1) / grid_x;
  const int nrg = (256 / cpb) / K;  // row-groups per block
  const long long rows_total = (long long)g.N * g.Ho;
  // aim for ~2048 blocks total to fill 256 CUs, but keep ≥~4 row-iterations
  // per thread so per-block atomics amortize
  long long chunks = std::max<long long>(1, kMaxGrid / grid_x);
  const long long by_iters = rows_total / std::max(1, nrg * 4);
  if (chunks > by_iters) chunks = by_iters;
  const long long max_chunks = (rows_total + nrg - 1) / std::max(1, nrg);
  if (chunks > max_chunks) chunks = max_chunks;
  if (chunks < 1) chunks = 1;
  const int rows_per_chunk = (int)((rows_total + chunks - 1) / chunks);
  chunks = (rows_total + rows_per_chunk - 1) / rows_per_chunk;
  dim3 grid(grid_x, (unsigned)chunks);
  const int lds = 256 * VEC * sizeof(float);
  const long long kkc = (long long)K * K * g.C;
  auto part = at::empty({chunks, (long long)K, (long long)K, (long long)g.C},
                        dw.options());
  dw_bwd_weight_kernel<T, K, VEC><<<grid, 256, lds, stream>>>(
      (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (float*)part.data_ptr(), g.N, g.C,
      g.H, g.W, g.Ho, g.Wo, g.sh, g.sw, g.ph, g.pw, cpb, rows_per_chunk);
  const int gx = dfd_grid(kkc, 256, 1 << 20);
  int zsplit = (int)std::min<long long>((2048 + gx - 1) / gx, (chunks + 7) / 8);
  if (zsplit < 1) zsplit = 1;
  const int z_per_blk = (int)((chunks + zsplit - 1) / zsplit);
  zsplit = (int)((chunks + z_per_blk - 1) / z_per_blk);
  if (zsplit > 1) dw.zero_();
  dw_bwd_weight_reduce_kernel<<<dim3(gx, zsplit), 256, 0, stream>>>(
      (const float*)part.data_ptr(), (float*)dw.data_ptr(), kkc, (int)chunks, z_per_blk);
}

template <typename T>
void bwd_weight_ktype(const at::Tensor& dy, const at::Tensor& x, at::Tensor& dw,
                      const Geom& g, int K, int vec, hipStream_t stream) {
  switch (K) {
#define DFD_CASE_K(KK)                                                      \
  case KK:                                                                  \
    switch (vec) {                                                          \
      case 8: launch_bwd_weight<T, KK, 8>(dy, x, dw, g, stream); break;     \
      case 4: launch_bwd_weight<T, KK, 4>(dy, x, dw, g, stream); break;     \
      case 2: launch_bwd_weight<T, KK, 2>(dy, x, dw, g, stream); break;     \
      default: launch_bwd_weight<T, KK, 1>(dy, x, dw, g, stream); break;    \
    }                                                                       \
    break;
    DFD_CASE_K(3)
    DFD_CASE_K(5)
    DFD_CASE_K(7)
    DFD_CASE_K(9)
    DFD_CASE_K(11)
#undef DFD_CASE_K
    default:
      TORCH_CHECK(false, "dwconv bwd_weight: unsupported kernel size ", K);
  }
}

Geom make_geom(int64_t N, int64_t C, int64_t H, int64_t W, int64_t K, int64_t sh,
               int64_t sw, int64_t ph, int64_t pw) {
  Geom g;
  g.N = (int)N;
  g.C = (int)C;
  g.H = (int)H;
  g.W = (int)W;
  g.sh = (int)sh;
  g.sw = (int)sw;
  g.ph = (int)ph;
  g.pw = (int)pw;
  g.Ho = (int)((H + 2 * ph - K) / sh + 1);
  g.Wo = (int)((W + 2 * pw - K) / sw + 1);
  return g;
}

}  // namespace

// x: NCHW logical, channels_last physical. w_packed: (K, K, C) same dtype.
template <typename T, int K>
void launch_fwd_s1_stats(const at::Tensor& x, const at::Tensor& w, at::Tensor& y,
                         at::Tensor& stats, const Geom& g, int vec, hipStream_t stream) {
  constexpr int TW = 4;
  const int wt = (g.Wo + TW - 1) / TW;
  const long long rows = (long long)g.N * g.Ho * wt;
  const int cv = g.C / vec;
  DfdPlan plan = dfd_plan(cv, rows);
  dim3 grid(plan.ctiles, plan.chunks);
  // fp32 stats scratch + this block's bf16/fp16/fp32 weight slice
  const int lds = 256 * vec * sizeof(float) +
                  K * K * plan.cpb * vec * (int)x.element_size();
#define DW_STATS_V(V)                                                          \
  dw_fwd_s1_stats_kernel<T, K, V, TW><<<grid, 256, lds, stream>>>(             \
      (const T*)x.data_ptr(), (const T*)w.data_ptr(), (T*)y.data_ptr(),        \
      stats.data_ptr<float>(), g.N, g.C, g.H, g.W, g.Ho, g.Wo, g.ph, g.pw,     \
      plan.cpb, plan.rows_per_chunk)
  switch (vec) {
    case 8: DW_STATS_V(8); break;
    case 4: DW_STATS_V(4); break;
    case 2: DW_STATS_V(2); break;
    default: DW_STATS_V(1); break;
  }
#undef DW_STATS_V
}

at::Tensor dw_conv2d_fwd(at::Tensor x, at::Tensor w_packed, int64_t sh, int64_t sw,
                         int64_t ph, int64_t pw, c10::optional<at::Tensor> stats_opt) {
  TORCH_CHECK(x.is_cuda() && w_packed.is_cuda(), "dwconv: CUDA tensors required");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "dwconv: x must be channels_last");
  TORCH_CHECK(w_packed.is_contiguous(), "dwconv: packed weight must be contiguous");
  const auto K = w_packed.size(0);
  TORCH_CHECK(w_packed.size(1) == K && w_packed.size(2) == x.size(1),
              "dwconv: packed weight shape mismatch");
  auto g = make_geom(x.size(0), x.size(1), x.size(2), x.size(3), K, sh, sw, ph, pw);
  auto y = at::empty({g.N, g.C, g.Ho, g.Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const int vec = pick_vec(g.C, (int)x.element_size());
  auto stream = at::hip::getCurrentHIPStream().stream();
  const auto stype = x.scalar_type();
  const int Ki = (int)K;
  if (stats_opt.has_value()) {
    at::Tensor stats = *stats_opt;
    TORCH_CHECK((Ki == 3 || Ki == 5) && sh == 1 && sw == 1,
                "dwconv: stats epilogue only for k3/k5 stride-1");
    TORCH_CHECK(stats.scalar_type() == at::kFloat && stats.is_contiguous() &&
                    stats.numel() == (long long)kDwStatsBuckets * 2 * g.C,
                "dwconv: stats must be fp32 [64, 2, C]");
#define DW_STATS_T(T)                                                          \
    do {                                                                       \
      if (Ki == 3)                                                             \
        launch_fwd_s1_stats<T, 3>(x, w_packed, y, stats, g, vec, stream);      \
      else                                                                     \
        launch_fwd_s1_stats<T, 5>(x, w_packed, y, stats, g, vec, stream);      \
    } while (0)
    switch (stype) {
      case at::kBFloat16: DW_STATS_T(__hip_bfloat16); break;
      case at::kHalf: DW_STATS_T(__half); break;
      case at::kFloat: DW_STATS_T(float); break;
      default: TORCH_CHECK(false, "dwconv: unsupported dtype");
    }
#undef DW_STATS_T
    return y;
  }
  switch (stype) {
    case at::kBFloat16: fwd_ktype<__hip_bfloat16>(x, w_packed, y, g, Ki, vec, stream); break;
    case at::kHalf: fwd_ktype<__half>(x, w_packed, y, g, Ki, vec, stream); break;
    case at::kFloat: fwd_ktype<float>(x, w_packed, y, g, Ki, vec, stream); break;
    default: TORCH_CHECK(false, "dwconv: unsupported dtype");
  }
  return y;
}

at::Tensor dw_conv2d_bwd_data(at::Tensor dy, at::Tensor w_packed, int64_t H, int64_t W,
                              int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  TORCH_CHECK(dy.is_cuda() && w_packed.is_cuda(), "dwconv: CUDA tensors required");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast), "dwconv: dy must be channels_last");
  const auto K = w_packed.size(0);
  Geom g = make_geom(dy.size(0), dy.size(1), H, W, K, sh, sw, ph, pw);
  TORCH_CHECK(g.Ho == dy.size(2) && g.Wo == dy.size(3), "dwconv bwd_data: geometry mismatch");
  auto dx = at::empty({g.N, g.C, g.H, g.W},
                      dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  const int vec = pick_vec(g.C, (int)dy.element_size());
  auto stream = at::hip::getCurrentHIPStream().stream();
  const auto stype = dy.scalar_type();
  const int Ki = (int)K;
  if (sh == 1 && sw == 1) {
    // stride-1 bwd-data is the same correlation with (kh,kw)-flipped weights
    // and padding K-1-p: run it through the TW-tiled forward kernel
    auto w_flip = at::flip(w_packed, {0, 1}).contiguous();
    Geom gf;
    gf.N = g.N; gf.C = g.C; gf.H = g.Ho; gf.W = g.Wo;
    gf.Ho = g.H; gf.Wo = g.W; gf.sh = 1; gf.sw = 1;
    gf.ph = Ki - 1 - g.ph; gf.pw = Ki - 1 - g.pw;
    switch (stype) {
      case at::kBFloat16: fwd_ktype<__hip_bfloat16>(dy, w_flip, dx, gf, Ki, vec, stream); break;
      case at::kHalf: fwd_ktype<__half>(dy, w_flip, dx, gf, Ki, vec, stream); break;
      case at::kFloat: fwd_ktype<float>(dy, w_flip, dx, gf, Ki, vec, stream); break;
      default: TORCH_CHECK(false, "dwconv: unsupported dtype");
    }
    return dx;
  }
  switch (stype) {
    case at::kBFloat16: bwd_data_ktype<__hip_bfloat16>(dy, w_packed, dx, g, Ki, vec, stream); break;
    case at::kHalf: bwd_data_ktype<__half>(dy, w_packed, dx, g, Ki, vec, stream); break;
    case at::kFloat: bwd_data_ktype<float>(dy, w_packed, dx, g, Ki, vec, stream); break;
    default: TORCH_CHECK(false, "dwconv: unsupported dtype");
  }
  return dx;
}

// returns (K, K, C) fp32
at::Tensor dw_conv2d_bwd_weight(at::Tensor dy, at::Tensor x, int64_t K, int64_t sh,
                                int64_t sw, int64_t ph, int64_t pw) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda(), "dwconv: CUDA tensors required");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "dwconv: x must be channels_last");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast), "dwconv: dy must be channels_last");
  Geom g = make_geom(x.size(0), x.size(1), x.size(2), x.size(3), K, sh, sw, ph, pw);
  TORCH_CHECK(g.Ho == dy.size(2) && g.Wo == dy.size(3), "dwconv bwd_weight: geometry mismatch");
  auto dw = at::empty({K, K, (long long)g.C}, x.options().dtype(at::kFloat));
  const int vec = pick_vec(g.C, (int)x.element_size());
  auto stream = at::hip::getCurrentHIPStream().stream();
  const auto stype = x.scalar_type();
  const int Ki = (int)K;
  switch (stype) {
    case at::kBFloat16: bwd_weight_ktype<__hip_bfloat16>(dy, x, dw, g, Ki, vec, stream); break;
    case at::kHalf: bwd_weight_ktype<__half>(dy, x, dw, g, Ki, vec, stream); break;
    case at::kFloat: bwd_weight_ktype<float>(dy, x, dw, g, Ki, vec, stream); break;
    default: TORCH_CHECK(false, "dwconv: unsupported dtype");
  }
  return dw;
}
