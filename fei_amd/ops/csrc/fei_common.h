// Common helpers for fei_amd gfx950 kernels.
// CDNA4 only: wave64, MFMA, LDS per /opt/skills/guides/cdna_hip_programming.md.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64

typedef __attribute__((ext_vector_type(8))) short s16x8;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef unsigned short u16;
typedef unsigned int u32;
typedef unsigned long long u64;

// bf16 <-> f32 at the bit level (bf16 = top 16 bits of f32; RNE rounding).
__device__ __forceinline__ float bf2f(u16 u) {
  union { float f; u32 i; } cv;
  cv.i = ((u32)u) << 16;
  return cv.f;
}
__device__ __forceinline__ u16 f2bf(float f) {
  union { float f; u32 i; } cv;
  cv.f = f;
  u32 x = cv.i;
  u32 lsb = (x >> 16) & 1u;
  x += 0x7fffu + lsb;                 // round to nearest even
  if ((cv.i & 0x7fffffffu) > 0x7f800000u) return (u16)((cv.i >> 16) | 0x0040u); // NaN
  return (u16)(x >> 16);
}

// dot of 8 bf16 pairs held as s16x8, accumulated in f32
__device__ __forceinline__ float dot8_bf16(s16x8 a, s16x8 b) {
  float acc = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i)
    acc = fmaf(bf2f((u16)a[i]), bf2f((u16)b[i]), acc);
  return acc;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

// Block reductions; `red` is LDS with >= blockDim.x/64 floats. All threads
// return the result. Two barriers protect `red` reuse across calls.
__device__ __forceinline__ float block_reduce_max(float v, float* red) {
  v = wave_reduce_max(v);
  const int nw = blockDim.x >> 6;
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) red[wid] = v;
  __syncthreads();
  v = red[0];
  for (int i = 1; i < nw; ++i) v = fmaxf(v, red[i]);
  __syncthreads();
  return v;
}
__device__ __forceinline__ float block_reduce_sum(float v, float* red) {
  v = wave_reduce_sum(v);
  const int nw = blockDim.x >> 6;
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) red[wid] = v;
  __syncthreads();
  v = red[0];
  for (int i = 1; i < nw; ++i) v += red[i];
  __syncthreads();
  return v;
}

// splitmix64 -> uniform float in (0, 1]; cheap per-element RNG for Gumbel.
__device__ __forceinline__ float hash_uniform(u64 x) {
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  x = x ^ (x >> 31);
  // take 24 high bits -> (0,1]
  u32 m = (u32)(x >> 40);
  return ((float)m + 1.0f) * (1.0f / 16777216.0f);
}

#define HIP_CHECK_LAUNCH() do { } while (0)
