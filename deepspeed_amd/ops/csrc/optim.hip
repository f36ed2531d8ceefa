// Fused Lion and LAMB optimizer kernels for MI355X (gfx950).
//
// Reference analogues: csrc/lion/multi_tensor_lion.cu and
// csrc/lamb/fused_lamb_cuda_kernel.cu. Same MI355X design stance as
// adam.hip: memory-bound grid-stride kernels over contiguous tensors,
// 16 B/lane vectorized fp32 streams, no multi-tensor-apply harness.
//
// LAMB runs as two passes per parameter tensor:
//   phase 1: Adam direction u = mhat/denom (+ wd*p), accumulating the
//            squared norms  sum(p^2) and sum(u^2) via wave+block
//            reduction and one atomicAdd per block (norms2[2]).
//   phase 2: trust ratio r = clamp(||p||/||u||, min, max) read from
//            norms2, p -= lr * r * u  (and optional bf16 mirror).

#include "ds_kernels.h"

namespace {

template <typename grad_t>
__global__ void fused_lion_kernel(float* __restrict__ p,
                                  const grad_t* __restrict__ g,
                                  float* __restrict__ m,
                                  ds::bf16* __restrict__ p16,
                                  const long long n, const float lr,
                                  const float beta1, const float beta2,
                                  const float weight_decay,
                                  const float inv_scale) {
  const long long stride = (long long)gridDim.x * blockDim.x * 4;
  long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  for (; base < n; base += stride) {
    const long long lim = min((long long)4, n - base);
    for (long long i = 0; i < lim; ++i) {
      const long long j = base + i;
      const float grad = ds::to_f32(g[j]) * inv_scale;
      const float c = beta1 * m[j] + (1.f - beta1) * grad;
      const float u = (c > 0.f ? 1.f : (c < 0.f ? -1.f : 0.f));
      p[j] -= lr * (u + weight_decay * p[j]);
      m[j] = beta2 * m[j] + (1.f - beta2) * grad;
      if (p16 != nullptr) p16[j] = ds::from_f32<ds::bf16>(p[j]);
    }
  }
}

template <typename grad_t>
__global__ void lamb_phase1_kernel(const float* __restrict__ p,
                                   const grad_t* __restrict__ g,
                                   float* __restrict__ m,
                                   float* __restrict__ v,
                                   float* __restrict__ u,
                                   float* __restrict__ norms2,  // [2]
                                   const long long n, const float beta1,
                                   const float beta2, const float eps,
                                   const float weight_decay,
                                   const float bias_corr1,
                                   const float bias_corr2_sqrt,
                                   const float inv_scale) {
  __shared__ float red[2][8];  // per-wave partials (<=8 waves)
  float acc_p = 0.f, acc_u = 0.f;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long j = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       j < n; j += stride) {
    const float grad = ds::to_f32(g[j]) * inv_scale;
    const float mi = beta1 * m[j] + (1.f - beta1) * grad;
    const float vi = beta2 * v[j] + (1.f - beta2) * grad * grad;
    m[j] = mi;
    v[j] = vi;
    const float mhat = mi / bias_corr1;
    const float denom = sqrtf(vi) / bias_corr2_sqrt + eps;
    float ui = mhat / denom + weight_decay * p[j];
    u[j] = ui;
    acc_p += p[j] * p[j];
    acc_u += ui * ui;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    acc_p += __shfl_down(acc_p, off, 64);
    acc_u += __shfl_down(acc_u, off, 64);
  }
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) {
    red[0][wave] = acc_p;
    red[1][wave] = acc_u;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float sp = 0.f, su = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) {
      sp += red[0][w];
      su += red[1][w];
    }
    atomicAdd(norms2 + 0, sp);
    atomicAdd(norms2 + 1, su);
  }
}

__global__ void lamb_phase2_kernel(float* __restrict__ p,
                                   const float* __restrict__ u,
                                   const float* __restrict__ norms2,
                                   ds::bf16* __restrict__ p16,
                                   const long long n, const float lr,
                                   const float max_coeff,
                                   const float min_coeff) {
  const float pn = sqrtf(norms2[0]);
  const float un = sqrtf(norms2[1]);
  float ratio = 1.f;
  if (pn > 0.f && un > 0.f)
    ratio = fminf(fmaxf(pn / un, min_coeff), max_coeff);
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long j = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       j < n; j += stride) {
    p[j] -= lr * ratio * u[j];
    if (p16 != nullptr) p16[j] = ds::from_f32<ds::bf16>(p[j]);
  }
}

}  // namespace

extern "C" void ds_fused_lion(float* p, const void* g, int grad_dtype,
                              float* m, void* p16, long long n, float lr,
                              float beta1, float beta2, float weight_decay,
                              float inv_scale, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const int block = 256;
  const int grid = ds::ds_num_blocks((n + 3) / 4, block);
  if (grad_dtype == 0)
    hipLaunchKernelGGL((fused_lion_kernel<float>), dim3(grid), dim3(block), 0,
                       st, p, (const float*)g, m, (ds::bf16*)p16, n, lr,
                       beta1, beta2, weight_decay, inv_scale);
  else if (grad_dtype == 1)
    hipLaunchKernelGGL((fused_lion_kernel<ds::bf16>), dim3(grid), dim3(block),
                       0, st, p, (const ds::bf16*)g, m, (ds::bf16*)p16, n, lr,
                       beta1, beta2, weight_decay, inv_scale);
  else
    hipLaunchKernelGGL((fused_lion_kernel<ds::f16>), dim3(grid), dim3(block),
                       0, st, p, (const ds::f16*)g, m, (ds::bf16*)p16, n, lr,
                       beta1, beta2, weight_decay, inv_scale);
}

extern "C" void ds_fused_lamb(float* p, const void* g, int grad_dtype,
                              float* m, float* v, float* u, float* norms2,
                              void* p16, long long n, float lr, float beta1,
                              float beta2, float eps, float weight_decay,
                              int step, float max_coeff, float min_coeff,
                              float inv_scale, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const float bc1 = 1.f - powf(beta1, (float)step);
  const float bc2s = sqrtf(1.f - powf(beta2, (float)step));
  const int block = 256;
  const int grid = ds::ds_num_blocks(n, block);
  hipMemsetAsync(norms2, 0, 2 * sizeof(float), st);
  if (grad_dtype == 0)
    hipLaunchKernelGGL((lamb_phase1_kernel<float>), dim3(grid), dim3(block),
                       0, st, p, (const float*)g, m, v, u, norms2, n, beta1,
                       beta2, eps, weight_decay, bc1, bc2s, inv_scale);
  else if (grad_dtype == 1)
    hipLaunchKernelGGL((lamb_phase1_kernel<ds::bf16>), dim3(grid),
                       dim3(block), 0, st, p, (const ds::bf16*)g, m, v, u,
                       norms2, n, beta1, beta2, eps, weight_decay, bc1, bc2s,
                       inv_scale);
  else
    hipLaunchKernelGGL((lamb_phase1_kernel<ds::f16>), dim3(grid), dim3(block),
                       0, st, p, (const ds::f16*)g, m, v, u, norms2, n, beta1,
                       beta2, eps, weight_decay, bc1, bc2s, inv_scale);
  hipLaunchKernelGGL(lamb_phase2_kernel, dim3(grid), dim3(block), 0, st, p, u,
                     norms2, (ds::bf16*)p16, n, lr, max_coeff, min_coeff);
}
