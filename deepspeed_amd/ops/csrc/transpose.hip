// Tiled bf16 matrix transpose for MI355X (gfx950).
//
// torch's permute(...).contiguous() runs the generic strided-copy kernel
// at ~250 GB/s on these shapes; this LDS-tiled version reads AND writes
// 16 B-coalesced (64x64 bf16 tiles, padded LDS) and is HBM-bound.
// Used by the flash-attention wrappers for the V^T / q^T / k^T / do^T
// layout changes (fwd vt and the bwd BHSD->BHDS copies).
//
// Batched: for batch index bh, the source matrix [R, C] starts at
//   src + (bh / inner) * outer_stride + (bh % inner) * inner_stride
// with row stride row_stride (elements). This covers both
//   [B, H, R, C] (inner=H, inner_stride=R*C... plain contiguous) and
//   [B, R, H, C] (inner=H, inner_stride=C, row_stride=H*C) layouts.
// Destination is contiguous [bh][C][R].

#include "ds_kernels.h"

namespace {

constexpr int TILE = 64;

__global__ void transpose_bf16_kernel(const short* __restrict__ src,
                                      short* __restrict__ dst,
                                      const int R, const int C,
                                      const long long row_stride,
                                      const int inner,
                                      const long long inner_stride,
                                      const long long outer_stride) {
  // +8 shorts padding per row: byte stride 136*... keeps the column reads
  // off a single bank without breaking 16 B alignment of rows
  __shared__ short tile[TILE][TILE + 8];

  const int bh = blockIdx.z;
  const long long sbase = (long long)(bh / inner) * outer_stride +
                          (long long)(bh % inner) * inner_stride;
  const long long dbase = (long long)bh * R * C;
  const int r0 = blockIdx.y * TILE;
  const int c0 = blockIdx.x * TILE;

  // 256 threads: read 64 rows x 64 cols, 16B per thread -> 8 rows/pass
  const int tid = threadIdx.x;
  const int rlane = tid & 7;          // 8 lanes x 8 shorts = 64 cols
  const int rrow = tid >> 3;          // 32 rows per pass
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int r = r0 + rrow + p * 32;
    if (r < R && c0 + rlane * 8 < C) {
      const short* s = src + sbase + (long long)r * row_stride + c0 +
                       rlane * 8;
      *(ds::bf16x8*)&tile[rrow + p * 32][rlane * 8] =
          *(const ds::bf16x8*)s;
    }
  }
  __syncthreads();

  // write transposed: thread writes 8 consecutive R-elements of one C-row
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int c = c0 + rrow + p * 32;           // output row = source col
    const int r = r0 + rlane * 8;               // output col = source row
    if (c < C && r < R) {
      ds::bf16x8 v;
#pragma unroll
      for (int i = 0; i < 8; ++i)
        v.v[i] = *(ds::bf16*)&tile[rlane * 8 + i][rrow + p * 32];
      // r is a multiple of 8 and R a multiple of 8 -> 16 B aligned store
      *(ds::bf16x8*)(dst + dbase + (long long)c * R + r) = v;
    }
  }
}

}  // namespace

extern "C" void ds_transpose_bf16(const void* src, void* dst, int n_batch,
                                  int R, int C, long long row_stride,
                                  int inner, long long inner_stride,
                                  long long outer_stride, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  dim3 grid((C + TILE - 1) / TILE, (R + TILE - 1) / TILE, n_batch);
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, st,
                     (const short*)src, (short*)dst, R, C, row_stride, inner,
                     inner_stride, outer_stride);
}
