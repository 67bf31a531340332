// Common device helpers for the distrl_llm_amd gfx950 kernel library.
// CDNA4-native: wave64, bf16x8 vector loads, LDS + cross-lane reductions.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cstdint>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// 16-byte vector of 8 bf16 values (coalescing sweet spot: 16 B/lane).
struct alignas(16) bf16x8 { __hip_bfloat16 v[8]; };
struct alignas(16) f32x4v { float v[4]; };

DEV_INLINE float bf2f(__hip_bfloat16 x) { return __bfloat162float(x); }
DEV_INLINE __hip_bfloat16 f2bf(float x) { return __float2bfloat16(x); }

// ---- cross-lane reductions (wave64) ----
DEV_INLINE float wave_sum(float x) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}
DEV_INLINE float wave_max(float x) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block reduction through LDS; `red` must have >= blockDim.x/WAVE slots.
DEV_INLINE float block_sum(float x, float* red) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  x = wave_sum(x);
  if (lane == 0) red[wid] = x;
  __syncthreads();
  float r = (threadIdx.x < nw) ? red[threadIdx.x] : 0.f;
  if (wid == 0) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) r += __shfl_xor(r, off, WAVE);
    if (lane == 0) red[0] = r;
  }
  __syncthreads();
  float out = red[0];
  __syncthreads();
  return out;
}

DEV_INLINE float block_max(float x, float* red) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  x = wave_max(x);
  if (lane == 0) red[wid] = x;
  __syncthreads();
  float r = (threadIdx.x < nw) ? red[threadIdx.x] : -1e30f;
  if (wid == 0) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) r = fmaxf(r, __shfl_xor(r, off, WAVE));
    if (lane == 0) red[0] = r;
  }
  __syncthreads();
  float out = red[0];
  __syncthreads();
  return out;
}

// ---- hash RNG (counter-based; deterministic per (seed, idx)) ----
DEV_INLINE uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
// uniform in (0, 1]
DEV_INLINE float hash_uniform(uint64_t seed, uint64_t idx) {
  uint64_t h = splitmix64(seed ^ splitmix64(idx));
  return ((h >> 40) + 1.0f) * (1.0f / 16777217.0f);
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e = hipGetLastError();                                        \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",               \
                hipGetErrorString(e));                                       \
  } while (0)

#define CDIV(a, b) (((a) + (b) - 1) / (b))
