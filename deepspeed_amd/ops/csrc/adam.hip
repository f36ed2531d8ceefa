// Fused Adam/AdamW for MI355X (gfx950).
//
// MI355X-first design: the ZeRO optimizers in this framework keep the
// fp32 master weights, optimizer state and 16-bit gradients as single
// contiguous flat shards, so the hot path is ONE memory-bound grid-stride
// kernel over flat buffers — no multi-tensor-apply chunk harness (the
// reference needs one because it steps thousands of separate tensors:
// csrc/adam/multi_tensor_adam.cu:129). Optionally writes the updated
// bf16 param shard in the same pass (saves the separate master->bf16 cast
// copy: one fewer full read+write of the shard at 8 TB/s HBM3E).
//
// Memory-bound: 4 streams read (p, g, m, v) + 3-4 written per element.
// Vectorized 16 B/lane (float4 on fp32 streams, bf16x8 nominal on grads);
// grad is read scalar-in-vector here because it shares the f32x4 loop
// width; the kernel is HBM-bound on the fp32 state traffic either way.

#include "ds_kernels.h"

namespace {

template <typename grad_t, bool ADAMW, bool WRITE_BF16>
__global__ void fused_adam_flat_kernel(
    float* __restrict__ p,        // fp32 master
    const grad_t* __restrict__ g, // gradient (bf16/fp16/fp32)
    float* __restrict__ m,
    float* __restrict__ v,
    ds::bf16* __restrict__ p16,   // optional bf16 mirror of p
    const long long n,
    const float lr,
    const float beta1,
    const float beta2,
    const float eps,
    const float weight_decay,
    const float bias_corr1,
    const float bias_corr2_sqrt,
    const float inv_scale) {
  const long long stride = (long long)gridDim.x * blockDim.x * 4;
  long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;

  for (; base < n; base += stride) {
    // vectorized 16B loads for fp32 streams when in-bounds
    if (base + 4 <= n) {
      ds::f32x4 pv = *reinterpret_cast<const ds::f32x4*>(p + base);
      ds::f32x4 mv = *reinterpret_cast<const ds::f32x4*>(m + base);
      ds::f32x4 vv = *reinterpret_cast<const ds::f32x4*>(v + base);
      float gv[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) gv[i] = ds::to_f32(g[base + i]) * inv_scale;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float grad = gv[i];
        if (!ADAMW && weight_decay != 0.f) grad += weight_decay * pv.v[i];
        mv.v[i] = beta1 * mv.v[i] + (1.f - beta1) * grad;
        vv.v[i] = beta2 * vv.v[i] + (1.f - beta2) * grad * grad;
        const float mhat = mv.v[i] / bias_corr1;
        const float denom = sqrtf(vv.v[i]) / bias_corr2_sqrt + eps;
        float update = mhat / denom;
        if (ADAMW && weight_decay != 0.f) update += weight_decay * pv.v[i];
        pv.v[i] -= lr * update;
      }
      *reinterpret_cast<ds::f32x4*>(p + base) = pv;
      *reinterpret_cast<ds::f32x4*>(m + base) = mv;
      *reinterpret_cast<ds::f32x4*>(v + base) = vv;
      if (WRITE_BF16) {
#pragma unroll
        for (int i = 0; i < 4; ++i) p16[base + i] = ds::from_f32<ds::bf16>(pv.v[i]);
      }
    } else {
      for (long long i = base; i < n; ++i) {
        float grad = ds::to_f32(g[i]) * inv_scale;
        if (!ADAMW && weight_decay != 0.f) grad += weight_decay * p[i];
        m[i] = beta1 * m[i] + (1.f - beta1) * grad;
        v[i] = beta2 * v[i] + (1.f - beta2) * grad * grad;
        const float mhat = m[i] / bias_corr1;
        const float denom = sqrtf(v[i]) / bias_corr2_sqrt + eps;
        float update = mhat / denom;
        if (ADAMW && weight_decay != 0.f) update += weight_decay * p[i];
        p[i] -= lr * update;
        if (WRITE_BF16) p16[i] = ds::from_f32<ds::bf16>(p[i]);
      }
    }
  }
}

template <typename grad_t>
void launch_typed(float* p, const grad_t* g, float* m, float* v,
                  ds::bf16* p16, long long n, float lr, float b1, float b2,
                  float eps, float wd, float bc1, float bc2s, float inv_scale,
                  bool adamw, hipStream_t stream) {
  const int block = 256;
  const int grid = ds::ds_num_blocks((n + 3) / 4, block);
  if (adamw) {
    if (p16)
      hipLaunchKernelGGL((fused_adam_flat_kernel<grad_t, true, true>), dim3(grid),
                         dim3(block), 0, stream, p, g, m, v, p16, n, lr, b1, b2,
                         eps, wd, bc1, bc2s, inv_scale);
    else
      hipLaunchKernelGGL((fused_adam_flat_kernel<grad_t, true, false>), dim3(grid),
                         dim3(block), 0, stream, p, g, m, v, p16, n, lr, b1, b2,
                         eps, wd, bc1, bc2s, inv_scale);
  } else {
    if (p16)
      hipLaunchKernelGGL((fused_adam_flat_kernel<grad_t, false, true>), dim3(grid),
                         dim3(block), 0, stream, p, g, m, v, p16, n, lr, b1, b2,
                         eps, wd, bc1, bc2s, inv_scale);
    else
      hipLaunchKernelGGL((fused_adam_flat_kernel<grad_t, false, false>), dim3(grid),
                         dim3(block), 0, stream, p, g, m, v, p16, n, lr, b1, b2,
                         eps, wd, bc1, bc2s, inv_scale);
  }
}

}  // namespace

extern "C" void ds_fused_adam_flat(float* p, const void* g, int grad_dtype,
                                   float* m, float* v, void* p16, long long n,
                                   float lr, float beta1, float beta2, float eps,
                                   float weight_decay, int step, float inv_scale,
                                   int adamw, void* stream) {
  const float bc1 = 1.f - powf(beta1, (float)step);
  const float bc2s = sqrtf(1.f - powf(beta2, (float)step));
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  ds::bf16* p16t = reinterpret_cast<ds::bf16*>(p16);
  switch (grad_dtype) {
    case 0:  // fp32
      launch_typed<float>(p, reinterpret_cast<const float*>(g), m, v, p16t, n, lr,
                          beta1, beta2, eps, weight_decay, bc1, bc2s, inv_scale,
                          adamw, s);
      break;
    case 1:  // bf16
      launch_typed<ds::bf16>(p, reinterpret_cast<const ds::bf16*>(g), m, v, p16t,
                             n, lr, beta1, beta2, eps, weight_decay, bc1, bc2s,
                             inv_scale, adamw, s);
      break;
    case 2:  // fp16
      launch_typed<ds::f16>(p, reinterpret_cast<const ds::f16*>(g), m, v, p16t, n,
                            lr, beta1, beta2, eps, weight_decay, bc1, bc2s,
                            inv_scale, adamw, s);
      break;
  }
}
