// Fused masked/alibi softmax and fused bias+dropout(+residual) kernels
// for MI355X (gfx950).
//
// Reference analogues: csrc/transformer/inference/csrc/softmax.cu:35
// (attn_softmax_v2 with mask/alibi variants) and
// csrc/transformer/dropout_kernels.cu (dropout + bias/residual fusions).
// CDNA4 design: one wave per row segment, wave64 shuffle reductions,
// 16 B vectorized loads; dropout uses a counter-based xorshift hash RNG
// (seed, element index) so masks are reproducible without rocRAND state.

#include "ds_kernels.h"

namespace {

// ---------------------------------------------------------------- softmax

// rows = B*H*Sq; row length n = Skv. Optional additive mask [B? broadcast]
// and alibi slope per head. x is bf16/fp16/fp32; compute fp32.
template <typename T>
__global__ void fused_softmax_kernel(const T* __restrict__ x,
                                     const T* __restrict__ mask,  // or null
                                     T* __restrict__ y,
                                     const float* __restrict__ alibi_slopes,
                                     const long long rows, const int n,
                                     const int heads, const int sq,
                                     const int mask_stride,  // elems per row
                                     const float scale, const int causal) {
  const long long row = (long long)blockIdx.x * (blockDim.x >> 6) +
                        (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const T* xr = x + row * n;
  T* yr = y + row * n;
  const int h = (int)((row / sq) % heads);
  const int qpos = (int)(row % sq);
  const float slope = alibi_slopes != nullptr ? alibi_slopes[h] : 0.f;
  const T* mr = mask != nullptr ? mask + (row / (heads * sq)) * mask_stride
                                : nullptr;
  const int limit = causal ? (n - sq + qpos + 1) : n;

  float m = -1e30f;
  for (int i = lane; i < limit; i += 64) {
    float v = ds::to_f32(xr[i]) * scale + slope * (float)(i - limit + 1);
    if (mr != nullptr) v += ds::to_f32(mr[i]);
    m = fmaxf(m, v);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    m = fmaxf(m, __shfl_xor(m, off, 64));
  float l = 0.f;
  for (int i = lane; i < limit; i += 64) {
    float v = ds::to_f32(xr[i]) * scale + slope * (float)(i - limit + 1);
    if (mr != nullptr) v += ds::to_f32(mr[i]);
    l += __expf(v - m);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    l += __shfl_xor(l, off, 64);
  const float inv = l > 0.f ? 1.f / l : 0.f;
  for (int i = lane; i < n; i += 64) {
    if (i >= limit) {
      yr[i] = ds::from_f32<T>(0.f);
    } else {
      float v = ds::to_f32(xr[i]) * scale + slope * (float)(i - limit + 1);
      if (mr != nullptr) v += ds::to_f32(mr[i]);
      yr[i] = ds::from_f32<T>(__expf(v - m) * inv);
    }
  }
}

// ---------------------------------------------------------------- dropout

__device__ __forceinline__ float hash_uniform(unsigned long long seed,
                                              unsigned long long idx) {
  unsigned long long z = seed ^ (idx * 0x9E3779B97F4A7C15ull);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  z = z ^ (z >> 31);
  return (float)(z >> 40) * (1.f / 16777216.f);  // 24-bit mantissa uniform
}

// y = dropout(x + bias) [+ residual]; mask output for the backward.
template <typename T, bool HAS_BIAS, bool HAS_RES>
__global__ void fused_dropout_kernel(const T* __restrict__ x,
                                     const T* __restrict__ bias,  // [cols]
                                     const T* __restrict__ residual,
                                     T* __restrict__ y,
                                     unsigned char* __restrict__ mask_out,
                                     const long long n, const int cols,
                                     const float ratio,
                                     const unsigned long long seed) {
  const float keep_scale = 1.f / (1.f - ratio);
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    float v = ds::to_f32(x[i]);
    if (HAS_BIAS) v += ds::to_f32(bias[i % cols]);
    const bool keep = hash_uniform(seed, (unsigned long long)i) >= ratio;
    v = keep ? v * keep_scale : 0.f;
    if (HAS_RES) v += ds::to_f32(residual[i]);
    y[i] = ds::from_f32<T>(v);
    mask_out[i] = keep ? 1 : 0;
  }
}

template <typename T>
__global__ void dropout_bwd_kernel(const T* __restrict__ dy,
                                   const unsigned char* __restrict__ mask,
                                   T* __restrict__ dx, const long long n,
                                   const float ratio) {
  const float keep_scale = 1.f / (1.f - ratio);
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride)
    dx[i] = ds::from_f32<T>(mask[i] ? ds::to_f32(dy[i]) * keep_scale : 0.f);
}

template <typename T>
void launch_sm(const void* x, const void* mask, void* y,
               const float* slopes, long long rows, int n, int heads, int sq,
               int mask_stride, float scale, int causal, hipStream_t st) {
  const int wpb = 4;
  const int grid = (int)((rows + wpb - 1) / wpb);
  hipLaunchKernelGGL((fused_softmax_kernel<T>), dim3(grid), dim3(wpb * 64),
                     0, st, (const T*)x, (const T*)mask, (T*)y, slopes, rows,
                     n, heads, sq, mask_stride, scale, causal);
}

template <typename T>
void launch_do(const void* x, const void* bias, const void* res, void* y,
               unsigned char* mask_out, long long n, int cols, float ratio,
               unsigned long long seed, hipStream_t st) {
  const int block = 256;
  const int grid = ds::ds_num_blocks(n, block);
#define GO(B_, R_) hipLaunchKernelGGL((fused_dropout_kernel<T, B_, R_>),      \
      dim3(grid), dim3(block), 0, st, (const T*)x, (const T*)bias,            \
      (const T*)res, (T*)y, mask_out, n, cols, ratio, seed)
  if (bias != nullptr && res != nullptr) GO(true, true);
  else if (bias != nullptr) GO(true, false);
  else if (res != nullptr) GO(false, true);
  else GO(false, false);
#undef GO
}

}  // namespace

extern "C" void ds_fused_softmax(const void* x, const void* mask, void* y,
                                 const float* alibi_slopes, long long rows,
                                 int n, int heads, int sq, int mask_stride,
                                 float scale, int causal, int dtype,
                                 void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (dtype == 0)
    launch_sm<float>(x, mask, y, alibi_slopes, rows, n, heads, sq,
                     mask_stride, scale, causal, st);
  else if (dtype == 1)
    launch_sm<ds::bf16>(x, mask, y, alibi_slopes, rows, n, heads, sq,
                        mask_stride, scale, causal, st);
  else
    launch_sm<ds::f16>(x, mask, y, alibi_slopes, rows, n, heads, sq,
                       mask_stride, scale, causal, st);
}

extern "C" void ds_fused_dropout(const void* x, const void* bias,
                                 const void* residual, void* y,
                                 unsigned char* mask_out, long long n,
                                 int cols, float ratio,
                                 unsigned long long seed, int dtype,
                                 void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (dtype == 0)
    launch_do<float>(x, bias, residual, y, mask_out, n, cols, ratio, seed, st);
  else if (dtype == 1)
    launch_do<ds::bf16>(x, bias, residual, y, mask_out, n, cols, ratio, seed,
                        st);
  else
    launch_do<ds::f16>(x, bias, residual, y, mask_out, n, cols, ratio, seed,
                       st);
}

extern "C" void ds_dropout_bwd(const void* dy, const unsigned char* mask,
                               void* dx, long long n, float ratio, int dtype,
                               void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const int block = 256;
  const int grid = ds::ds_num_blocks(n, block);
  if (dtype == 0)
    hipLaunchKernelGGL((dropout_bwd_kernel<float>), dim3(grid), dim3(block),
                       0, st, (const float*)dy, mask, (float*)dx, n, ratio);
  else if (dtype == 1)
    hipLaunchKernelGGL((dropout_bwd_kernel<ds::bf16>), dim3(grid),
                       dim3(block), 0, st, (const ds::bf16*)dy, mask,
                       (ds::bf16*)dx, n, ratio);
  else
    hipLaunchKernelGGL((dropout_bwd_kernel<ds::f16>), dim3(grid), dim3(block),
                       0, st, (const ds::f16*)dy, mask, (ds::f16*)dx, n,
                       ratio);
}
