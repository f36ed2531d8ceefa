// Rotary position embedding (RoPE) for MI355X (gfx950).
//
// Capability parity with the reference's apply_rotary_pos_emb.cu
// (csrc/transformer/inference/csrc/apply_rotary_pos_emb.cu:27, neox
// rotate-half style, GQA-aware), re-designed for CDNA4:
//  * cos/sin are a precomputed fp32 host table [max_seq, D/2] (per the CDNA4
//    guide, on-device sinf/cosf turns this memory-bound op VALU-bound)
//  * layout [B, S, Hh, D] (the natural attention-projection output; avoids
//    the reference's 0213 transpose kernels entirely)
//  * one thread per rotation pair, grid-stride; backward = inverse rotation
//    (negated sin), same kernel.

#include "ds_kernels.h"

namespace {

template <typename T, bool BWD>
__global__ void rope_kernel(T* __restrict__ x,        // in-place [B,S,Hh,D]
                            const float* __restrict__ cs,  // [max_seq, D/2] cos
                            const float* __restrict__ sn,  // [max_seq, D/2] sin
                            const int* __restrict__ positions,  // [B,S] or null
                            const long long total_pairs,
                            const int S, const int Hh, const int D) {
  const int half = D / 2;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total_pairs; i += stride) {
    const int j = (int)(i % half);
    const long long row = i / half;        // (b*S + s)*Hh + h
    const long long bs = row / Hh;         // b*S + s
    const int s = (int)(bs % S);
    const int pos = positions ? positions[bs] : s;
    const float c = cs[(long long)pos * half + j];
    const float sv = BWD ? -sn[(long long)pos * half + j]
                         : sn[(long long)pos * half + j];
    T* base = x + row * D;
    const float x1 = ds::to_f32(base[j]);
    const float x2 = ds::to_f32(base[j + half]);
    base[j] = ds::from_f32<T>(x1 * c - x2 * sv);
    base[j + half] = ds::from_f32<T>(x2 * c + x1 * sv);
  }
}

template <typename T>
void launch_rope(void* x, const float* cs, const float* sn,
                 const int* positions, long long total_pairs, int S, int Hh,
                 int D, bool bwd, hipStream_t s) {
  const int block = 256;
  const int grid = ds::ds_num_blocks(total_pairs, block);
  if (bwd)
    hipLaunchKernelGGL((rope_kernel<T, true>), dim3(grid), dim3(block), 0, s,
                       reinterpret_cast<T*>(x), cs, sn, positions, total_pairs,
                       S, Hh, D);
  else
    hipLaunchKernelGGL((rope_kernel<T, false>), dim3(grid), dim3(block), 0, s,
                       reinterpret_cast<T*>(x), cs, sn, positions, total_pairs,
                       S, Hh, D);
}

}  // namespace

extern "C" void ds_rope(void* x, const float* cos_table, const float* sin_table,
                        const int* positions, long long batch, long long seq,
                        long long heads, long long dim, int bwd, int dtype,
                        void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  const long long total_pairs = batch * seq * heads * (dim / 2);
  if (dtype == 1)
    launch_rope<ds::bf16>(x, cos_table, sin_table, positions, total_pairs,
                          (int)seq, (int)heads, (int)dim, bwd != 0, s);
  else if (dtype == 2)
    launch_rope<ds::f16>(x, cos_table, sin_table, positions, total_pairs,
                         (int)seq, (int)heads, (int)dim, bwd != 0, s);
  else
    launch_rope<float>(x, cos_table, sin_table, positions, total_pairs,
                       (int)seq, (int)heads, (int)dim, bwd != 0, s);
}
