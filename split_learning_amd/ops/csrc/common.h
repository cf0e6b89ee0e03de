// Common helpers for the split_learning_amd CDNA4 (gfx950) kernels.
// All kernels here are written directly in HIP for MI355X — wave64, MFMA
// matrix cores, LDS — no CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define SLK_WAVE 64

using f32x2 = __attribute__((ext_vector_type(2))) float;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;

#define HIP_CHECK(expr)                                                          \
  do {                                                                           \
    hipError_t _e = (expr);                                                      \
    if (_e != hipSuccess) {                                                      \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ", __FILE__, \
                  ":", __LINE__);                                                \
    }                                                                            \
  } while (0)

static inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

// Zero-fill as a KERNEL node, never hipMemsetAsync: captured memset nodes were
// observed to intermittently misorder against dependent kernel nodes on
// hipGraph replay (per-process at graphExec instantiation; root-caused by
// bisection — see profiles/SUMMARY.md "graph divergence").  Kernel nodes
// order correctly; cost is the same single launch.
static __global__ void slk_zero_kernel(float* __restrict__ p, long n) {
  // float4 main body + scalar tail (torch allocations are 16B-aligned)
  const long n4 = n >> 2;
  f32x4* __restrict__ p4 = reinterpret_cast<f32x4*>(p);
  const long stride = (long)gridDim.x * blockDim.x;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long i = i0; i < n4; i += stride) p4[i] = f32x4{};
  for (long i = (n4 << 2) + i0; i < n; i += stride) p[i] = 0.f;
}

static inline void slk_zero_async(float* p, long n, hipStream_t stream) {
  const int grid = (int)std::min<long>((n / 4 + 255) / 256 + 1, 4096);
  hipLaunchKernelGGL(slk_zero_kernel, dim3(grid), dim3(256), 0, stream, p, n);
}

// splitmix64 — counter-based RNG hash for dropout (deterministic per
// (seed, offset, index); quality is ample for Bernoulli masks).
__device__ __forceinline__ uint64_t slk_mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

__device__ __forceinline__ float slk_uniform(uint64_t seed, uint64_t offset,
                                             uint64_t idx) {
  uint64_t h = slk_mix64(seed ^ slk_mix64(offset * 0xD1342543DE82EF95ull + idx));
  // top 24 bits -> [0, 1)
  return (float)(h >> 40) * (1.0f / 16777216.0f);
}

// 32-bit mask hash (lowbias32) for dropout draws INSIDE compute kernels:
// ~6 VALU vs ~25 for the 64-bit path (two splitmix64 = four 64-bit
// multiplies) — the difference made the fused attention kernel VALU-bound
// at BERT scale.  Fold (seed, offset) into one 32-bit stream id host/SALU-
// side via slk_mix64 first.
__device__ __forceinline__ float slk_uniform32(uint32_t stream, uint32_t idx) {
  uint32_t x = idx * 0x9E3779B9u + stream;
  x ^= x >> 16;
  x *= 0x7FEB352Du;
  x ^= x >> 15;
  x *= 0x846CA68Bu;
  x ^= x >> 16;
  return (float)(x >> 8) * (1.0f / 16777216.0f);
}

// Fast integer division by a runtime constant (one 64-bit mul + shift).
// Valid for dividend < 2^24 and divisor < 2^16 (all tensor-index math here):
// m = floor(2^40/d)+1; q = (n*m) >> 40 == n/d exactly when n*d < 2^40.
struct FastDiv {
  unsigned d;
  unsigned long long m;
  void init(unsigned d_) {
    d = d_ == 0 ? 1 : d_;
    m = (0x10000000000ull / d) + 1;
  }
  __device__ __forceinline__ unsigned div(unsigned n) const {
    return (unsigned)(((unsigned long long)n * m) >> 40);
  }
  __device__ __forceinline__ unsigned mod(unsigned n, unsigned q) const {
    return n - q * d;
  }
};

// block-level reduction into lane 0 of wave 0 (sums `val` over blockDim.x
// threads; blockDim.x must be a multiple of 64 and <= 1024)
template <typename T>
__device__ __forceinline__ T slk_block_sum(T val, T* lds_scratch /* >= 16 */) {
  // wave reduce
  for (int off = 32; off > 0; off >>= 1) val += __shfl_down(val, off, 64);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int nwaves = blockDim.x >> 6;
  if (lane == 0) lds_scratch[wid] = val;
  __syncthreads();
  T out = (T)0;
  if (wid == 0) {
    out = (lane < nwaves) ? lds_scratch[lane] : (T)0;
    for (int off = 8; off > 0; off >>= 1) out += __shfl_down(out, off, 64);
  }
  return out;  // valid in thread 0 only
}
