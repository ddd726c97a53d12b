// Common device helpers for NornicDB-AMD CDNA4 (gfx950) kernels.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; all cross-lane idioms use width-64 shuffles.
//  - memory-bound kernels vectorize bf16 loads as short4/short8 (8-16B/lane).
//  - no FP32-input MFMA on CDNA4; fp32 paths use the vector ALU.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// Vector types for wide loads/stores.
typedef short short4v __attribute__((ext_vector_type(4)));    // 8B: 4 x bf16
typedef short short8v __attribute__((ext_vector_type(8)));    // 16B: 8 x bf16
typedef float float4v __attribute__((ext_vector_type(4)));    // 16B
typedef float float2v __attribute__((ext_vector_type(2)));
typedef unsigned int uint4v __attribute__((ext_vector_type(4)));

DEV_INLINE float bf16_bits_to_f32(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

DEV_INLINE unsigned short f32_to_bf16_bits(float f) {
  union { unsigned int i; float f; } v;
  v.f = f;
  // round-to-nearest-even
  unsigned int lsb = (v.i >> 16) & 1u;
  v.i += 0x7fffu + lsb;
  return (unsigned short)(v.i >> 16);
}

// Full-wave (64-lane) reductions via xor shuffles.
DEV_INLINE float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

DEV_INLINE float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block reduction: sums one float per lane across the whole block.
// `scratch` must hold >= blockDim.x/WAVE floats.
DEV_INLINE float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  x = (threadIdx.x < nw) ? scratch[threadIdx.x] : 0.0f;
  if (wid == 0) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  }
  if (threadIdx.x == 0) scratch[0] = x;
  __syncthreads();
  float r = scratch[0];
  __syncthreads();
  return r;
}

// splitmix64 — cheap deterministic per-element hash for synthetic data gen.
DEV_INLINE uint64_t splitmix64(uint64_t z) {
  z += 0x9e3779b97f4a7c15ULL;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ULL;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebULL;
  return z ^ (z >> 31);
}

// Two approximately-N(0,1) floats from one 64-bit hash (Box-Muller-lite:
// sum of 4 uniforms - 2, variance 1/3 each -> scale sqrt(3)). Cheap, no trig;
// good enough for synthetic benchmark corpora.
DEV_INLINE float2v hash_gauss2(uint64_t key) {
  uint64_t h = splitmix64(key);
  uint32_t a = (uint32_t)h, b = (uint32_t)(h >> 32);
  uint64_t h2 = splitmix64(key ^ 0xdeadbeefcafef00dULL);
  uint32_t c = (uint32_t)h2, d = (uint32_t)(h2 >> 32);
  const float s = 1.0f / 4294967296.0f;
  float u0 = (float)a * s, u1 = (float)b * s, u2 = (float)c * s, u3 = (float)d * s;
  float2v r;
  r.x = (u0 + u1 + u2 + u3 - 2.0f) * 1.7320508f;
  uint64_t h3 = splitmix64(key ^ 0x123456789abcdefULL);
  uint32_t e = (uint32_t)h3, f = (uint32_t)(h3 >> 32);
  float u4 = (float)e * s, u5 = (float)f * s;
  r.y = (u0 - u1 + u4 - u5) * 1.7320508f;
  return r;
}

#define HIP_CHECK_LAST()                                                    \
  do {                                                                      \
    hipError_t _e = hipGetLastError();                                      \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(_e)); \
    }                                                                       \
  } while (0)
