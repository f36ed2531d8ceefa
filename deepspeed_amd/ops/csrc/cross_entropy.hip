// Fused cross-entropy for MI355X (gfx950) — bf16 logits -> loss + dlogits
// without materializing fp32 logits.
//
// Reference analogue: the training loss path materializes [tokens, vocab]
// fp32 logits (2 GB per 4k tokens at V=128k) and runs softmax + nll as
// separate kernels; this computes the log-sum-exp reduction and the
// backward scatter directly from bf16, fp32 in-register.
//
// fwd: one 256-thread workgroup per token row; grid-stride over V.
//   pass 1: row max (wave shuffle + LDS reduce)
//   pass 2: sum exp(x - max); writes lse[i] and loss_i = lse - x[y_i]
// bwd: elementwise grid-stride over [N, V]:
//   dlogit[i,j] = (exp(x - lse_i) - (j == y_i)) * gscale_i
// Ignore-index rows contribute loss 0 and zero gradients.

#include "ds_kernels.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8s;
constexpr int CT = 256;

__device__ __forceinline__ float bf2f(short s) {
  union {
    float f;
    unsigned u;
  } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return c.f;
}

__device__ __forceinline__ float block_max_256(float x, float* lds) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  x = ds::wave_reduce_max(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  float t = lds[0];
#pragma unroll
  for (int i = 1; i < CT / 64; ++i) t = fmaxf(t, lds[i]);
  __syncthreads();
  return t;
}

__device__ __forceinline__ float block_sum_256(float x, float* lds) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  x = ds::wave_reduce_sum(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  float t = 0.f;
#pragma unroll
  for (int i = 0; i < CT / 64; ++i) t += lds[i];
  __syncthreads();
  return t;
}

__global__ __launch_bounds__(CT) void ce_fwd_kernel(
    const short* __restrict__ logits,  // [N, V] bf16
    const long long* __restrict__ labels,  // [N]
    float* __restrict__ loss,          // [N]
    float* __restrict__ lse,           // [N]
    const int N, const long long V, const long long ignore_index) {
  __shared__ float lds[CT / 64];
  const int row = blockIdx.x;
  if (row >= N) return;
  const short* x = logits + (long long)row * V;
  const long long y = labels[row];

  float m = -1e30f;
  for (long long j = threadIdx.x; j < V; j += CT)
    m = fmaxf(m, bf2f(x[j]));
  m = block_max_256(m, lds);

  float s = 0.f;
  for (long long j = threadIdx.x; j < V; j += CT)
    s += __expf(bf2f(x[j]) - m);
  s = block_sum_256(s, lds);

  if (threadIdx.x == 0) {
    const float l = m + __logf(s);
    lse[row] = l;
    loss[row] = (y == ignore_index) ? 0.f : l - bf2f(x[y]);
  }
}

__global__ void ce_bwd_kernel(
    const short* __restrict__ logits, const long long* __restrict__ labels,
    const float* __restrict__ lse, const float* __restrict__ gscale,  // [N]
    short* __restrict__ dlogits, const int N, const long long V,
    const long long ignore_index) {
  const long long total = (long long)N * V;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += stride) {
    const long long row = i / V, j = i % V;
    const long long y = labels[row];
    float g;
    if (y == ignore_index) {
      g = 0.f;
    } else {
      g = (__expf(bf2f(logits[i]) - lse[row]) - (j == y ? 1.f : 0.f)) *
          gscale[row];
    }
    dlogits[i] = (short)__bfloat16_as_ushort(__float2bfloat16(g));
  }
}

}  // namespace

extern "C" void ds_ce_fwd(const void* logits, const long long* labels,
                          float* loss, float* lse, int N, long long V,
                          long long ignore_index, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(N), dim3(CT), 0, st,
                     (const short*)logits, labels, loss, lse, N, V,
                     ignore_index);
}

extern "C" void ds_ce_bwd(const void* logits, const long long* labels,
                          const float* lse, const float* gscale,
                          void* dlogits, int N, long long V,
                          long long ignore_index, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const int grid = ds::ds_num_blocks((long long)N * V / 4, 256);
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(grid), dim3(256), 0, st,
                     (const short*)logits, labels, lse, gscale,
                     (short*)dlogits, N, V, ignore_index);
}
