// Fused SwiGLU (SiLU-gated MLP activation) + bias-GELU for MI355X (gfx950).
//
// Capability parity with the reference's gated-activation kernels
// (inference/v2/kernels/core_ops/gated_activations/, csrc gelu.cu), fused
// fwd+bwd for training. Memory-bound; 16 B/lane vectorized; grid-stride.

#include "ds_kernels.h"

namespace {

__device__ __forceinline__ float silu(float x) {
  return x / (1.f + expf(-x));
}
__device__ __forceinline__ float silu_grad(float x) {
  const float sig = 1.f / (1.f + expf(-x));
  return sig * (1.f + x * (1.f - sig));
}
__device__ __forceinline__ float gelu_tanh(float x) {
  const float k = 0.7978845608028654f;  // sqrt(2/pi)
  return 0.5f * x * (1.f + tanhf(k * (x + 0.044715f * x * x * x)));
}
__device__ __forceinline__ float gelu_tanh_grad(float x) {
  const float k = 0.7978845608028654f;
  const float x3 = x * x * x;
  const float t = tanhf(k * (x + 0.044715f * x3));
  const float dt = (1.f - t * t) * k * (1.f + 3.f * 0.044715f * x * x);
  return 0.5f * (1.f + t) + 0.5f * x * dt;
}

template <typename T, int ACT>  // ACT: 0=silu-gate, 1=gelu-gate
__global__ void gated_act_fwd_kernel(const T* __restrict__ gate,
                                     const T* __restrict__ up,
                                     T* __restrict__ out, const long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       base < n; base += stride) {
    if (base + 8 <= n && sizeof(T) == 2) {
      T g[8], u[8], o[8];
      *reinterpret_cast<ds::bf16x8*>(g) =
          *reinterpret_cast<const ds::bf16x8*>(gate + base);
      *reinterpret_cast<ds::bf16x8*>(u) =
          *reinterpret_cast<const ds::bf16x8*>(up + base);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float gv = ds::to_f32(g[i]);
        const float act = ACT == 0 ? silu(gv) : gelu_tanh(gv);
        o[i] = ds::from_f32<T>(act * ds::to_f32(u[i]));
      }
      *reinterpret_cast<ds::bf16x8*>(out + base) =
          *reinterpret_cast<const ds::bf16x8*>(o);
    } else {
      for (long long i = base; i < n && i < base + 8; ++i) {
        const float gv = ds::to_f32(gate[i]);
        const float act = ACT == 0 ? silu(gv) : gelu_tanh(gv);
        out[i] = ds::from_f32<T>(act * ds::to_f32(up[i]));
      }
    }
  }
}

template <typename T, int ACT>
__global__ void gated_act_bwd_kernel(const T* __restrict__ dout,
                                     const T* __restrict__ gate,
                                     const T* __restrict__ up,
                                     T* __restrict__ dgate,
                                     T* __restrict__ dup, const long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float dov = ds::to_f32(dout[i]);
    const float gv = ds::to_f32(gate[i]);
    const float uv = ds::to_f32(up[i]);
    const float act = ACT == 0 ? silu(gv) : gelu_tanh(gv);
    const float dact = ACT == 0 ? silu_grad(gv) : gelu_tanh_grad(gv);
    dgate[i] = ds::from_f32<T>(dov * uv * dact);
    dup[i] = ds::from_f32<T>(dov * act);
  }
}

}  // namespace

extern "C" void ds_gated_act_fwd(const void* gate, const void* up, void* out,
                                 long long n, int act, int dtype, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  const int block = 256;
  const int grid = ds::ds_num_blocks((n + 7) / 8, block);
#define DISPATCH(T)                                                          \
  if (act == 0)                                                              \
    hipLaunchKernelGGL((gated_act_fwd_kernel<T, 0>), dim3(grid), dim3(block), \
                       0, s, reinterpret_cast<const T*>(gate),               \
                       reinterpret_cast<const T*>(up),                       \
                       reinterpret_cast<T*>(out), n);                        \
  else                                                                       \
    hipLaunchKernelGGL((gated_act_fwd_kernel<T, 1>), dim3(grid), dim3(block), \
                       0, s, reinterpret_cast<const T*>(gate),               \
                       reinterpret_cast<const T*>(up),                       \
                       reinterpret_cast<T*>(out), n)
  if (dtype == 1) { DISPATCH(ds::bf16); }
  else if (dtype == 2) { DISPATCH(ds::f16); }
  else { DISPATCH(float); }
#undef DISPATCH
}

extern "C" void ds_gated_act_bwd(const void* dout, const void* gate,
                                 const void* up, void* dgate, void* dup,
                                 long long n, int act, int dtype, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  const int block = 256;
  const int grid = ds::ds_num_blocks(n, block);
#define DISPATCH(T)                                                           \
  if (act == 0)                                                               \
    hipLaunchKernelGGL((gated_act_bwd_kernel<T, 0>), dim3(grid), dim3(block),  \
                       0, s, reinterpret_cast<const T*>(dout),                \
                       reinterpret_cast<const T*>(gate),                      \
                       reinterpret_cast<const T*>(up),                        \
                       reinterpret_cast<T*>(dgate), reinterpret_cast<T*>(dup), \
                       n);                                                    \
  else                                                                        \
    hipLaunchKernelGGL((gated_act_bwd_kernel<T, 1>), dim3(grid), dim3(block),  \
                       0, s, reinterpret_cast<const T*>(dout),                \
                       reinterpret_cast<const T*>(gate),                      \
                       reinterpret_cast<const T*>(up),                        \
                       reinterpret_cast<T*>(dgate), reinterpret_cast<T*>(dup), \
                       n)
  if (dtype == 1) { DISPATCH(ds::bf16); }
  else if (dtype == 2) { DISPATCH(ds::f16); }
  else { DISPATCH(float); }
#undef DISPATCH
}
