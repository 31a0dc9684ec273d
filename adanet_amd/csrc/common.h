// Common device helpers for adanet_amd CDNA4 (gfx950) kernels.
//
// MI355X-native conventions used throughout csrc/:
//   * wavefront = 64 lanes (CDNA), never 32.
//   * bf16 compute with fp32 accumulation (MFMA f32_16x16x32_bf16).
//   * memory-bound kernels load bf16 vectorized as short4/short8.
//   * grid-stride loops capped near 2048 blocks (256 CUs x 8 blocks).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define WAVE 64

using bf16_t = __hip_bfloat16;

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(2))) short s16x2;
typedef __attribute__((ext_vector_type(4))) int i32x4;

__device__ __forceinline__ float bf2f(bf16_t x) { return __bfloat162float(x); }
__device__ __forceinline__ bf16_t f2bf(float x) { return __float2bfloat16(x); }

__device__ __forceinline__ float bits2f(short s) {
  union { unsigned int u; float f; } cvt;
  cvt.u = ((unsigned int)(unsigned short)s) << 16;
  return cvt.f;
}

__device__ __forceinline__ short f2bits(float f) {
  union { unsigned int u; float f; } cvt;
  cvt.f = f;
  // round-to-nearest-even bf16 truncation
  unsigned int u = cvt.u;
  unsigned int rounding = 0x7fff + ((u >> 16) & 1);
  return (short)((u + rounding) >> 16);
}

// Wave-level (64-lane) reductions via xor shuffles.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Stateless counter-based RNG (pcg-like hash) for dropout: reproducible from
// (seed, index) so backward can recompute the mask without storing it.
__device__ __forceinline__ uint32_t hash_rng(uint64_t seed, uint64_t idx) {
  uint64_t z = seed + 0x9e3779b97f4a7c15ull * (idx + 1);
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return (uint32_t)(z >> 32);
}

#define HIP_CHECK_KERNEL()                                                    \
  do {                                                                        \
    hipError_t e = hipGetLastError();                                         \
    if (e != hipSuccess) {                                                    \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
    }                                                                         \
  } while (0)
