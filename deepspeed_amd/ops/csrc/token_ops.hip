// Token gather/scatter for random-LTD on MI355X (gfx950).
//
// Reference analogue: csrc/random_ltd/gather_scatter.cu (gather_tokens /
// scatter_tokens) + token_sort.cu. Re-designed: one wave per selected
// token row, 16 B per lane, grid-stride over (batch, token) pairs — a
// [B,S,D] hidden-state row move is pure HBM bandwidth, so the kernel is
// just maximally-coalesced row copies driven by an index list. The
// torch.gather path materializes an expanded [B,K,D] int64 index tensor
// (8 bytes of index traffic per 2-byte element); this kernel reads the
// [B,K] indices once per row.

#include "ds_kernels.h"

namespace {

// y[b, k, :] = x[b, idx[b, k], :]      (GATHER == 1)
// y[b, idx[b, k], :] = x[b, k, :]      (GATHER == 0; y preloaded with the
//                                       passthrough sequence)
template <typename T, int GATHER>
__global__ void token_move_kernel(const T* __restrict__ x,
                                  T* __restrict__ y,
                                  const int* __restrict__ idx,
                                  const int B, const int S, const int K,
                                  const int D) {
  const int waves_per_block = blockDim.x >> 6;
  const int lane = threadIdx.x & 63;
  const int vec = 16 / sizeof(T);
  for (long long row = (long long)blockIdx.x * waves_per_block +
                       (threadIdx.x >> 6);
       row < (long long)B * K;
       row += (long long)gridDim.x * waves_per_block) {
    const int b = (int)(row / K);
    const int k = (int)(row % K);
    const int s = idx[b * K + k];
    const T* src;
    T* dst;
    if (GATHER) {
      src = x + ((long long)b * S + s) * D;
      dst = y + ((long long)b * K + k) * D;
    } else {
      src = x + ((long long)b * K + k) * D;
      dst = y + ((long long)b * S + s) * D;
    }
    for (int d = lane * vec; d < D; d += 64 * vec) {
      *reinterpret_cast<float4*>(dst + d) =
          *reinterpret_cast<const float4*>(src + d);
    }
  }
}

template <typename T>
void launch(const void* x, void* y, const int* idx, int B, int S, int K,
            int D, int gather, hipStream_t st) {
  const int waves_per_block = 4;
  const int block = waves_per_block * 64;
  long long rows = (long long)B * K;
  long long want = (rows + waves_per_block - 1) / waves_per_block;
  if (want < 1) want = 1;
  const int grid = (int)(want > 4096 ? 4096 : want);
  if (gather)
    hipLaunchKernelGGL((token_move_kernel<T, 1>), dim3(grid), dim3(block), 0,
                       st, (const T*)x, (T*)y, idx, B, S, K, D);
  else
    hipLaunchKernelGGL((token_move_kernel<T, 0>), dim3(grid), dim3(block), 0,
                       st, (const T*)x, (T*)y, idx, B, S, K, D);
}

}  // namespace

extern "C" void ds_token_move(const void* x, void* y, const int* idx, int B,
                              int S, int K, int D, int gather, int dtype,
                              void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (dtype == 0)
    launch<float>(x, y, idx, B, S, K, D, gather, st);
  else if (dtype == 1)
    launch<ds::bf16>(x, y, idx, B, S, K, D, gather, st);
  else
    launch<ds::f16>(x, y, idx, B, S, K, D, gather, st);
}
