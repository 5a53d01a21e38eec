// Common device utilities for the MI355X (gfx950/CDNA4) kernels.
// Wave width is 64 on CDNA4; every cross-lane idiom below assumes it.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#include <cstdint>

#define WAVE_SIZE 64

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",      \
                  __FILE__, ":", __LINE__);                                 \
    }                                                                       \
  } while (0)

namespace bpa {

template <typename T>
__host__ __device__ __forceinline__ T tmin(T a, T b) { return a < b ? a : b; }

// column-reduce partials chunk (col_reduce_kernel grid.y granularity)
constexpr int kColChunk = 16;

// ---------------------------------------------------------------------------
// dtype conversion helpers (bf16/fp16/fp32 <-> fp32 compute)
// ---------------------------------------------------------------------------
template <typename T>
struct DTraits;

template <>
struct DTraits<float> {
  static __device__ __forceinline__ float to_f32(float v) { return v; }
  static __device__ __forceinline__ float from_f32(float v) { return v; }
};

template <>
struct DTraits<__hip_bfloat16> {
  static __device__ __forceinline__ float to_f32(__hip_bfloat16 v) {
    return __bfloat162float(v);
  }
  static __device__ __forceinline__ __hip_bfloat16 from_f32(float v) {
    return __float2bfloat16(v);
  }
};

template <>
struct DTraits<__half> {
  static __device__ __forceinline__ float to_f32(__half v) {
    return __half2float(v);
  }
  static __device__ __forceinline__ __half from_f32(float v) {
    return __float2half(v);
  }
};

// ---------------------------------------------------------------------------
// wave-level reductions (64 lanes)
// ---------------------------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// cross-wave sum of an (a,b) pair through LDS; result broadcast to all
// threads. smem must hold 2 * (blockDim.x/64) floats. Single-use per
// __shared__ buffer (no trailing barrier).
__device__ __forceinline__ void block_reduce_sum2(float& a, float& b,
                                                  float* smem) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  a = wave_reduce_sum(a);
  b = wave_reduce_sum(b);
  if (nwaves == 1) return;
  if (lane == 0) {
    smem[2 * wave] = a;
    smem[2 * wave + 1] = b;
  }
  __syncthreads();
  a = 0.f;
  b = 0.f;
  for (int w = 0; w < nwaves; ++w) {
    a += smem[2 * w];
    b += smem[2 * w + 1];
  }
}

// reduce across a block of NW waves through LDS; result broadcast to all.
template <int NW>
__device__ __forceinline__ float block_reduce_sum(float v, float* smem) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) smem[wave] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < NW; ++w) total += smem[w];
  return total;
}

// ---------------------------------------------------------------------------
// Philox4x32-10 counter-based RNG (for fused dropout). (Measured: the
// round count is NOT what makes the dropout path expensive - 7 vs 10
// rounds was wall-time neutral on the attention kernels - so the
// standard 10-round variant is kept.)
// ---------------------------------------------------------------------------
struct Philox {
  uint32_t k0, k1;
  __device__ Philox(uint64_t seed) {
    k0 = static_cast<uint32_t>(seed);
    k1 = static_cast<uint32_t>(seed >> 32);
  }
  static __device__ __forceinline__ uint32_t mulhi(uint32_t a, uint32_t b) {
    return static_cast<uint32_t>((static_cast<uint64_t>(a) * b) >> 32);
  }
  // returns 4 uniform u32 for counter `ctr`
  __device__ __forceinline__ void operator()(uint64_t ctr, uint32_t out[4]) const {
    uint32_t c0 = static_cast<uint32_t>(ctr);
    uint32_t c1 = static_cast<uint32_t>(ctr >> 32);
    uint32_t c2 = 0, c3 = 0;
    uint32_t key0 = k0, key1 = k1;
    constexpr uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
    constexpr uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
    for (int round = 0; round < 10; ++round) {
      uint32_t hi0 = mulhi(M0, c0), lo0 = M0 * c0;
      uint32_t hi1 = mulhi(M1, c2), lo1 = M1 * c2;
      uint32_t n0 = hi1 ^ c1 ^ key0;
      uint32_t n1 = lo1;
      uint32_t n2 = hi0 ^ c3 ^ key1;
      uint32_t n3 = lo0;
      c0 = n0; c1 = n1; c2 = n2; c3 = n3;
      key0 += W0; key1 += W1;
    }
    out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
  }
};

// uniform in [0,1) from u32
__device__ __forceinline__ float u32_to_uniform(uint32_t x) {
  return static_cast<float>(x) * (1.0f / 4294967296.0f);
}

// exact-erf GELU and its derivative (matches the fp32 torch reference)
__device__ __forceinline__ float gelu_fwd(float x) {
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752440f));
}

__device__ __forceinline__ float gelu_bwd(float x) {
  // d/dx [x * Phi(x)] = Phi(x) + x * phi(x)
  const float kInvSqrt2 = 0.70710678118654752440f;
  const float kInvSqrt2Pi = 0.39894228040143267794f;
  float cdf = 0.5f * (1.0f + erff(x * kInvSqrt2));
  float pdf = kInvSqrt2Pi * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}

}  // namespace bpa
