// Common helpers for the gfx950 (CDNA4) kernels.
//
// Conventions (per /opt/skills/guides/cdna_hip_programming.md):
//   * wavefront = 64 lanes; blocks are multiples of 64
//   * bf16/fp16 global access vectorized (>=8 B/lane)
//   * memory-bound ops grid-stride with grid capped near 256 CU x 8 blocks
//   * fp32 accumulation everywhere (bf16 storage)
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define DFD_DEV __device__ __forceinline__

constexpr int kWave = 64;
constexpr int kMaxGrid = 2048;  // ~256 CUs x 8 blocks

static inline int dfd_grid(long long total, int block, int cap = kMaxGrid) {
  long long g = (total + block - 1) / block;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return (int)g;
}

// ---- scalar conversions ---------------------------------------------------
template <typename T> struct DfdCvt;

template <> struct DfdCvt<float> {
  static DFD_DEV float to_f32(float v) { return v; }
  static DFD_DEV float from_f32(float v) { return v; }
};
template <> struct DfdCvt<__hip_bfloat16> {
  static DFD_DEV float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }
  static DFD_DEV __hip_bfloat16 from_f32(float v) { return __float2bfloat16(v); }
};
template <> struct DfdCvt<__half> {
  static DFD_DEV float to_f32(__half v) { return __half2float(v); }
  static DFD_DEV __half from_f32(float v) { return __float2half(v); }
};

// ---- activations ----------------------------------------------------------
enum class Act : int { kNone = 0, kRelu = 1, kSilu = 2 };

DFD_DEV float act_fwd(float z, Act act) {
  switch (act) {
    case Act::kRelu: return z > 0.f ? z : 0.f;
    case Act::kSilu: {
      const float s = 1.f / (1.f + __expf(-z));
      return z * s;
    }
    default: return z;
  }
}

// d(act)/dz evaluated at pre-activation z
DFD_DEV float act_bwd(float z, Act act) {
  switch (act) {
    case Act::kRelu: return z > 0.f ? 1.f : 0.f;
    case Act::kSilu: {
      const float s = 1.f / (1.f + __expf(-z));
      return s * (1.f + z * (1.f - s));
    }
    default: return 1.f;
  }
}

// ---- wave/block reductions ------------------------------------------------
DFD_DEV float wave_sum(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  return v;
}

// block-wide sum over `nwaves` waves via LDS; result valid on thread 0
template <int kMaxWaves = 16>
DFD_DEV float block_sum(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  v = wave_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  const int nwaves = (blockDim.x + kWave - 1) / kWave;
  float out = 0.f;
  if (wid == 0) {
    out = (lane < nwaves) ? lds_scratch[lane] : 0.f;
    out = wave_sum(out);
  }
  __syncthreads();
  return out;
}

// ---- vectorized channel slices -------------------------------------------
// Every NHWC kernel moves VEC contiguous channels per thread (16 B loads for
// 2-byte dtypes at VEC=8).
template <typename T, int N>
struct alignas(sizeof(T) * N) DfdVec {
  T v[N];
};

template <typename T, int N>
DFD_DEV DfdVec<T, N> dfd_vload(const T* p) {
  return *reinterpret_cast<const DfdVec<T, N>*>(p);
}

template <typename T, int N>
DFD_DEV void dfd_vstore(T* p, const DfdVec<T, N>& x) {
  *reinterpret_cast<DfdVec<T, N>*>(p) = x;
}

// largest power-of-two vector width (≤16 B per lane) dividing C
static inline int dfd_pick_vec(long long c, int elem_size) {
  const int max_vec = elem_size == 4 ? 4 : 8;
  for (int v = max_vec; v > 1; v >>= 1)
    if (c % v == 0) return v;
  return 1;
}

// ---- (channel-slot, row-group) block plan --------------------------------
// Blocks are cpb power-of-two channel slots × row groups; the grid splits
// rows into chunks. Chunks aim for ~kMaxGrid blocks but keep ≥min_iters
// row-iterations per thread so per-block reductions/atomics amortize.
struct DfdPlan {
  int cpb, ctiles, chunks, rows_per_chunk;
};

// Channel slots are NOT rounded up to a power of two (that idled up to 44%
// of a block's load bandwidth on C=144-class layers); tiles balance.
static inline DfdPlan dfd_plan(int cv, long long rows, long long base_blocks = 1,
                               int min_iters = 16) {
  DfdPlan p;
  p.ctiles = (cv + 63) / 64;
  p.cpb = (cv + p.ctiles - 1) / p.ctiles;
  const long long base = base_blocks * p.ctiles;
  long long want = (kMaxGrid + base - 1) / base;
  const int nrg = 256 / p.cpb;
  long long by_iters = rows / ((long long)nrg * min_iters);
  if (want > by_iters) want = by_iters;
  long long max_chunks = (rows + nrg - 1) / nrg;
  if (want > max_chunks) want = max_chunks;
  if (want < 1) want = 1;
  p.chunks = (int)want;
  p.rows_per_chunk = (int)((rows + p.chunks - 1) / p.chunks);
  return p;
}
