// prime_amd — common device helpers for CDNA4 (gfx950 / MI355X) kernels.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; block sizes are multiples of 64.
//  - memory-bound kernels vectorize bf16 loads as short4/short8 reinterprets
//    (hipcc does not auto-vectorize scalar bf16 loads) and use grid-stride
//    loops capped near 256 CUs * 8 blocks.
//  - fp32 accumulation everywhere; bf16 is storage format only.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define PRIME_WAVE 64
#define PRIME_MAX_GRID 2048  // ~256 CU * 8 blocks

typedef __hip_bfloat16 bf16;

// ---- vector types for wide loads --------------------------------------
struct alignas(16) bf16x8 { bf16 v[8]; };   // 16 B / lane
struct alignas(8)  bf16x4 { bf16 v[4]; };   // 8 B / lane
struct alignas(16) f32x4v { float v[4]; };

__device__ __forceinline__ float bf2f(bf16 x) { return __bfloat162float(x); }
__device__ __forceinline__ bf16 f2bf(float x) { return __float2bfloat16(x); }

// ---- wave + block reductions ------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, 64);
  return x;  // valid in lane 0 of the wave
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_down(x, off, 64));
  return x;
}

// Block-level sum reduction over up to 1024 threads; returns result on ALL
// threads. `scratch` must be LDS with >= blockDim.x/64 floats.
__device__ __forceinline__ float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = (blockDim.x + 63) >> 6;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float r = 0.f;
  if (threadIdx.x < (unsigned)nwaves) r = scratch[threadIdx.x];
  // reduce the (<=16) partials in wave 0 and broadcast via LDS
  if (wid == 0) {
    r = wave_reduce_sum(r);
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  r = scratch[0];
  __syncthreads();
  return r;
}

__device__ __forceinline__ float block_reduce_max(float x, float* scratch) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = (blockDim.x + 63) >> 6;
  x = wave_reduce_max(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float r = -INFINITY;
  if (threadIdx.x < (unsigned)nwaves) r = scratch[threadIdx.x];
  if (wid == 0) {
    r = wave_reduce_max(r);
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  r = scratch[0];
  __syncthreads();
  return r;
}

static inline int prime_grid(int64_t work_items, int block) {
  int64_t g = (work_items + block - 1) / block;
  if (g > PRIME_MAX_GRID) g = PRIME_MAX_GRID;
  if (g < 1) g = 1;
  return (int)g;
}

#define PRIME_API extern "C" __attribute__((visibility("default")))
