// Shared helpers for the deepspeed_amd CDNA4 (gfx950 / MI355X) kernels.
//
// Written directly for CDNA4: wavefront = 64 lanes, 32-bank x 4B LDS,
// HBM3E-bound elementwise ops vectorized to 16 B/lane.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

namespace ds {

using bf16 = __hip_bfloat16;
using f16 = __half;

// 16-byte vector of 8 bf16 values (the coalescing sweet spot on CDNA4).
struct alignas(16) bf16x8 {
  bf16 v[8];
};
struct alignas(16) f32x4 {
  float v[4];
};

__device__ __forceinline__ float to_f32(bf16 x) { return __bfloat162float(x); }
__device__ __forceinline__ float to_f32(f16 x) { return __half2float(x); }
__device__ __forceinline__ float to_f32(float x) { return x; }

template <typename T>
__device__ __forceinline__ T from_f32(float x);
template <>
__device__ __forceinline__ bf16 from_f32<bf16>(float x) {
  return __float2bfloat16(x);
}
template <>
__device__ __forceinline__ f16 from_f32<f16>(float x) {
  return __float2half(x);
}
template <>
__device__ __forceinline__ float from_f32<float>(float x) {
  return x;
}

// Full-wave (64-lane) reductions via xor shuffles.
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  return x;
}

// Block-level reduction: one partial per wave staged through LDS; every
// thread returns the block total (LDS broadcast reads are conflict-free).
// BLOCK must be a multiple of 64 and <= 1024. Safe to call repeatedly with
// the same scratch (trailing barrier).
template <int BLOCK>
__device__ __forceinline__ float block_reduce_sum(float x, float* lds_scratch) {
  constexpr int NWAVES = BLOCK / WAVE_SIZE;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds_scratch[wid] = x;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < NWAVES; ++i) total += lds_scratch[i];
  __syncthreads();
  return total;
}

// Grid sizing for memory-bound grid-stride kernels: cap around 2048 blocks
// (256 CUs x 8 blocks) per the CDNA4 guide, grid-stride the rest.
inline int ds_num_blocks(long long work_items, int block) {
  long long blocks = (work_items + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return static_cast<int>(blocks);
}

}  // namespace ds
