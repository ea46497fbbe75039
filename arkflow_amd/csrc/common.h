// Shared device helpers for arkflow_amd gfx950 kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

DEV_INLINE uint64_t lanemask_lt() {
  return (1ull << (threadIdx.x & 63)) - 1ull;
}

// 64-bit splittable mix (Stafford variant 13) — key hashing.
DEV_INLINE uint64_t mix64(uint64_t z) {
  z ^= z >> 30; z *= 0xbf58476d1ce4e5b9ull;
  z ^= z >> 27; z *= 0x94d049bb133111ebull;
  z ^= z >> 31;
  return z;
}

// wave-wide sum reduce (f32)
DEV_INLINE float wave_reduce_sum(float v) {
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}
DEV_INLINE float wave_reduce_max(float v) {
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// atomic float max/min via monotone flip to UNSIGNED int order
// (the flipped values must be compared unsigned — top-bit-set patterns)
DEV_INLINE uint32_t float_flip(float f) {
  uint32_t u = (uint32_t)__float_as_int(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
DEV_INLINE float float_unflip(uint32_t u) {
  return __int_as_float(
      (int32_t)((u & 0x80000000u) ? (u & 0x7fffffffu) : ~u));
}

constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }
