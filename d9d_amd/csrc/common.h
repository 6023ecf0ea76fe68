// Common device helpers for d9d_amd CDNA4 (gfx950) kernels.
//
// Wave64 throughout: CDNA wavefront is 64 lanes (not 32). Block sizes are
// multiples of 64. bf16 memory traffic is vectorized as ushort4/ushort8
// (8/16 B per lane) — hipcc does not auto-vectorize scalar bf16 loads.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define D9D_DEVICE __device__ __forceinline__

namespace d9d {

constexpr int kWaveSize = 64;

// ---- vector types ----------------------------------------------------------

typedef ushort __attribute__((ext_vector_type(4))) ushort4v;
typedef ushort __attribute__((ext_vector_type(8))) ushort8v;
typedef float __attribute__((ext_vector_type(4))) float4v;

union Bf16x8 {
  ushort8v u;
  ushort s[8];
};

D9D_DEVICE float bf16_bits_to_f32(ushort bits) {
  union {
    uint32_t u;
    float f;
  } cvt;
  cvt.u = static_cast<uint32_t>(bits) << 16;
  return cvt.f;
}

// Round-to-nearest-even f32 -> bf16 bits.
D9D_DEVICE ushort f32_to_bf16_rne(float x) {
  union {
    float f;
    uint32_t u;
  } cvt;
  cvt.f = x;
  uint32_t u = cvt.u;
  if ((u & 0x7fffffffu) > 0x7f800000u) {  // NaN
    return static_cast<ushort>((u >> 16) | 0x0040u);
  }
  uint32_t rounding_bias = 0x7fffu + ((u >> 16) & 1u);
  return static_cast<ushort>((u + rounding_bias) >> 16);
}

// Stochastic-rounding f32 -> bf16 bits given 16 bits of noise.
D9D_DEVICE ushort f32_to_bf16_stochastic(float x, uint32_t noise16) {
  union {
    float f;
    uint32_t u;
  } cvt;
  cvt.f = x;
  uint32_t u = cvt.u;
  if ((u & 0x7fffffffu) > 0x7f800000u) {  // NaN
    return static_cast<ushort>((u >> 16) | 0x0040u);
  }
  return static_cast<ushort>((u + (noise16 & 0xffffu)) >> 16);
}

// ---- RNG: counter-based splitmix64 (per-element, stateless) ----------------

D9D_DEVICE uint64_t splitmix64(uint64_t z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

// ---- reductions -------------------------------------------------------------

// Sum over all 64 lanes of the wave; every lane gets the result.
D9D_DEVICE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v += __shfl_xor(v, off, kWaveSize);
  }
  return v;
}

D9D_DEVICE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_xor(v, off, kWaveSize));
  }
  return v;
}

// Block-level sum over up to 16 waves via LDS scratch (caller provides
// a float[16] shared buffer). Every thread gets the result.
template <int BLOCK>
D9D_DEVICE float block_reduce_sum(float v, float* lds_scratch) {
  static_assert(BLOCK % kWaveSize == 0, "block must be whole waves");
  constexpr int kWaves = BLOCK / kWaveSize;
  v = wave_reduce_sum(v);
  if constexpr (kWaves == 1) {
    return v;
  }
  const int wave = threadIdx.x / kWaveSize;
  const int lane = threadIdx.x % kWaveSize;
  if (lane == 0) lds_scratch[wave] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < kWaves; ++w) total += lds_scratch[w];
  __syncthreads();
  return total;
}

__host__ __device__ __forceinline__ int64_t ceil_div(int64_t a, int64_t b) {
  return (a + b - 1) / b;
}

}  // namespace d9d
