// Groupwise symmetric int8/int4 quantization for MI355X (gfx950).
//
// Capability parity with the reference's QuantizerBuilder kernels
// (csrc/quantization/quantize.cu:23 cached_quantization,
// dequantize.cu:12) — new CDNA4 design: ONE 256-thread workgroup per
// quantization group; the absmax reduction uses 64-lane xor-shuffle waves
// + a 4-entry LDS stage, and every global access is coalesced along the
// flat buffer. Used by ZeRO++-style quantized weight all-gather (qwZ):
// bf16 shard -> int8 + fp32 group scales, RCCL ships half the bytes over
// xGMI, consumer dequantizes into the bf16 full buffer.
//
// int4 packs two values per byte (lo nibble = even index).

#include "ds_kernels.h"

namespace {

constexpr int QTHREADS = 256;

__device__ __forceinline__ float block_reduce_max_256(float x) {
  __shared__ float lds[QTHREADS / WAVE_SIZE];
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  x = ds::wave_reduce_max(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  float total = lds[0];
#pragma unroll
  for (int i = 1; i < QTHREADS / WAVE_SIZE; ++i) total = fmaxf(total, lds[i]);
  __syncthreads();
  return total;
}

template <typename T, int BITS>
__global__ void groupwise_quant_kernel(const T* __restrict__ x,
                                       int8_t* __restrict__ q,
                                       float* __restrict__ scales,
                                       const long long n,
                                       const int group_size) {
  const long long g = blockIdx.x;  // one workgroup per group
  const long long start = g * group_size;
  const long long end = min(start + (long long)group_size, n);

  float amax = 0.f;
  for (long long i = start + threadIdx.x; i < end; i += QTHREADS)
    amax = fmaxf(amax, fabsf(ds::to_f32(x[i])));
  amax = block_reduce_max_256(amax);

  const float qmax = BITS == 8 ? 127.f : 7.f;
  const float scale = amax > 0.f ? amax / qmax : 1.f;
  const float inv = 1.f / scale;
  if (threadIdx.x == 0) scales[g] = scale;

  if (BITS == 8) {
    for (long long i = start + threadIdx.x; i < end; i += QTHREADS) {
      const float v = ds::to_f32(x[i]) * inv;
      q[i] = (int8_t)__float2int_rn(fmaxf(fminf(v, 127.f), -127.f));
    }
  } else {
    // int4: threads own byte pairs (2 elements each)
    for (long long b = start / 2 + threadIdx.x; b * 2 + 1 < end + (end & 1);
         b += QTHREADS) {
      const long long i0 = b * 2, i1 = b * 2 + 1;
      if (i0 < start || i0 >= end) continue;
      float v0 = ds::to_f32(x[i0]) * inv;
      int q0 = __float2int_rn(fmaxf(fminf(v0, 7.f), -7.f)) & 0xF;
      int q1 = 0;
      if (i1 < end) {
        float v1 = ds::to_f32(x[i1]) * inv;
        q1 = __float2int_rn(fmaxf(fminf(v1, 7.f), -7.f)) & 0xF;
      }
      q[b] = (int8_t)((q1 << 4) | q0);
    }
  }
}

template <typename T, int BITS>
__global__ void groupwise_dequant_kernel(const int8_t* __restrict__ q,
                                         const float* __restrict__ scales,
                                         T* __restrict__ out,
                                         const long long n,
                                         const int group_size) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float scale = scales[i / group_size];
    float v;
    if (BITS == 8) {
      v = (float)q[i] * scale;
    } else {
      int byte = q[i / 2];
      int nib = (i & 1) ? ((byte >> 4) & 0xF) : (byte & 0xF);
      if (nib & 0x8) nib -= 16;  // sign-extend 4-bit
      v = (float)nib * scale;
    }
    out[i] = ds::from_f32<T>(v);
  }
}

template <typename T>
void launch_quant(const T* x, int8_t* q, float* s, long long n, int gs,
                  int bits, hipStream_t st) {
  const long long groups = (n + gs - 1) / gs;
  if (bits == 8)
    hipLaunchKernelGGL((groupwise_quant_kernel<T, 8>), dim3(groups),
                       dim3(QTHREADS), 0, st, x, q, s, n, gs);
  else
    hipLaunchKernelGGL((groupwise_quant_kernel<T, 4>), dim3(groups),
                       dim3(QTHREADS), 0, st, x, q, s, n, gs);
}

template <typename T>
void launch_dequant(const int8_t* q, const float* s, T* out, long long n,
                    int gs, int bits, hipStream_t st) {
  const int block = 256;
  const int grid = ds::ds_num_blocks(n, block);
  if (bits == 8)
    hipLaunchKernelGGL((groupwise_dequant_kernel<T, 8>), dim3(grid),
                       dim3(block), 0, st, q, s, out, n, gs);
  else
    hipLaunchKernelGGL((groupwise_dequant_kernel<T, 4>), dim3(grid),
                       dim3(block), 0, st, q, s, out, n, gs);
}

}  // namespace

extern "C" void ds_groupwise_quant(const void* x, int dtype, void* q,
                                   float* scales, long long n, int group_size,
                                   int bits, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  int8_t* qp = reinterpret_cast<int8_t*>(q);
  switch (dtype) {
    case 0:
      launch_quant(reinterpret_cast<const float*>(x), qp, scales, n,
                   group_size, bits, st);
      break;
    case 1:
      launch_quant(reinterpret_cast<const ds::bf16*>(x), qp, scales, n,
                   group_size, bits, st);
      break;
    case 2:
      launch_quant(reinterpret_cast<const ds::f16*>(x), qp, scales, n,
                   group_size, bits, st);
      break;
  }
}

extern "C" void ds_groupwise_dequant(const void* q, const float* scales,
                                     void* out, int dtype, long long n,
                                     int group_size, int bits, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const int8_t* qp = reinterpret_cast<const int8_t*>(q);
  switch (dtype) {
    case 0:
      launch_dequant(qp, scales, reinterpret_cast<float*>(out), n, group_size,
                     bits, st);
      break;
    case 1:
      launch_dequant(qp, scales, reinterpret_cast<ds::bf16*>(out), n,
                     group_size, bits, st);
      break;
    case 2:
      launch_dequant(qp, scales, reinterpret_cast<ds::f16*>(out), n,
                     group_size, bits, st);
      break;
  }
}
