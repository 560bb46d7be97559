// Shared helpers for rllm_amd CDNA4 (gfx950) kernels.
//
// MI355X-first conventions used throughout (see /opt/skills guides):
//  * wave = 64 lanes; block sizes are multiples of 64.
//  * bf16 global loads are vectorized (short4/short8 reinterpret) -- hipcc
//    does not auto-vectorize scalar bf16 loads.
//  * reductions: wave-level __shfl_xor tree, then LDS across waves.
//  * memory-bound kernels grid-stride with grid capped near 2048 blocks.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE_SIZE 64

using bf16 = __hip_bfloat16;

// Vector types for wide loads.
typedef short  short4v  __attribute__((ext_vector_type(4)));
typedef short  short8v  __attribute__((ext_vector_type(8)));
typedef float  float4v  __attribute__((ext_vector_type(4)));
typedef float  float2v  __attribute__((ext_vector_type(2)));

__device__ __forceinline__ float bf16_to_f32(uint16_t u) {
  union { uint32_t u32; float f; } cvt;
  cvt.u32 = ((uint32_t)u) << 16;
  return cvt.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
  union { float f; uint32_t u32; } cvt;
  cvt.f = f;
  uint32_t u = cvt.u32;
  // round-to-nearest-even
  uint32_t rounding = 0x7fff + ((u >> 16) & 1);
  u += rounding;
  return (uint16_t)(u >> 16);
}

// ---- wave-level reductions (64 lanes) ----
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// ---- block-level reductions (<= 16 waves / 1024 threads) ----
// `scratch` must be __shared__ float[16]. Result valid in all threads.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : 0.f;
  // nwaves <= 16, reduce within the first wave then broadcast via LDS
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) r += __shfl_xor(r, off, WAVE_SIZE);
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  return scratch[0];
}

__device__ __forceinline__ float block_reduce_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : -INFINITY;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) r = fmaxf(r, __shfl_xor(r, off, WAVE_SIZE));
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  return scratch[0];
}

// ---- RNG: counter-based hash -> uniform(0,1] -> gumbel ----
// Philox-lite: enough statistical quality for categorical sampling.
__device__ __forceinline__ uint32_t hash_u32(uint32_t a, uint32_t b, uint32_t c) {
  uint32_t h = a * 0x9E3779B9u ^ b * 0x85EBCA6Bu ^ c * 0xC2B2AE35u;
  h ^= h >> 16; h *= 0x7FEB352Du;
  h ^= h >> 15; h *= 0x846CA68Bu;
  h ^= h >> 16;
  return h;
}

__device__ __forceinline__ float uniform_from_u32(uint32_t u) {
  // (0, 1]: avoids log(0) in gumbel transform
  return ((float)(u >> 8) + 1.0f) * (1.0f / 16777216.0f);
}

// Grid sizing for memory-bound grid-stride kernels (guide §6 G11).
static inline int grid_for(int64_t total_threads_needed, int block) {
  int64_t blocks = (total_threads_needed + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

#define HIP_CHECK_KERNEL() do { \
  hipError_t e = hipGetLastError(); \
  if (e != hipSuccess) { \
    TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
  } \
} while (0)
