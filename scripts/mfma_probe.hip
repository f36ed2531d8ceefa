// Empirical MFMA fragment-layout probe for gfx950 (MI355X).
//
// Identifies, for mfma_f32_32x32x16_bf16 and mfma_f32_16x16x32_bf16, the
// complete (lane, reg) -> (row, k) mapping of operand A and
// (lane, reg) -> (col, k) of operand B, with k as a consistent class label
// (absolute k order is irrelevant to GEMM correctness — A and B just have
// to agree).
//
// Method: for every A slot (64 lanes x 8 regs = 512), run
// D = mfma(onehot_A, valueB, 0) twice, where valueB encodes each B slot's
// lane (pass 1) or reg (pass 2) as an exact-bf16 integer. The nonzero row
// of D is row(A slot); D[row][col] equals the encoded id of the unique B
// slot sharing A's k at that col. Host groups A slots by their B-slot
// signature to assign k classes.
//
// Build: hipcc --offload-arch=gfx950 -O2 -o mfma_probe mfma_probe.hip
// Output: CSV per shape: tag, row, then 32 (or 16) b-slot ids per col
//         for pass1/pass2 interleaved as blane,breg pairs.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <vector>

typedef __attribute__((ext_vector_type(8))) short bf16x8s;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef float f32x16 __attribute__((ext_vector_type(16)));

__device__ inline short bf(float x) {
  return (short)__bfloat16_as_ushort(__float2bfloat16(x));
}

// out layout per tag: [row, b_id[0..NC-1]] with b_id = blane*16 + breg
// (encoded from the two passes inside the kernel)
template <int NROW, int NCOL>
__global__ void probe(int* out) {
  const int lane = threadIdx.x;
  __shared__ float d1s[NROW * NCOL];
  __shared__ float d2s[NROW * NCOL];
  for (int tag = 0; tag < 512; ++tag) {
    const int alane = tag >> 3, areg = tag & 7;
    bf16x8s a = {};
    bf16x8s b1, b2;
    for (int i = 0; i < 8; ++i) {
      b1[i] = bf((float)(lane + 1));
      b2[i] = bf((float)(i + 1));
    }
    if (lane == alane) a[areg] = bf(1.0f);

    if (NROW == 32) {
      f32x16 d1 = {}, d2 = {};
      d1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b1, d1, 0, 0, 0);
      d2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b2, d2, 0, 0, 0);
      for (int r = 0; r < 16; ++r) {
        const int col = lane & 31;
        const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        d1s[row * NCOL + col] = d1[r];
        d2s[row * NCOL + col] = d2[r];
      }
    } else {
      f32x4 d1 = {}, d2 = {};
      d1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, d1, 0, 0, 0);
      d2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b2, d2, 0, 0, 0);
      for (int r = 0; r < 4; ++r) {
        const int col = lane & 15;
        const int row = (lane >> 4) * 4 + r;
        d1s[row * NCOL + col] = d1[r];
        d2s[row * NCOL + col] = d2[r];
      }
    }
    __syncthreads();
    if (lane == 0) {
      int row_hit = -1;
      for (int r = 0; r < NROW && row_hit < 0; ++r)
        for (int c = 0; c < NCOL; ++c)
          if (d1s[r * NCOL + c] > 0.5f) {
            row_hit = r;
            break;
          }
      out[tag * (1 + NCOL)] = row_hit;
      for (int c = 0; c < NCOL; ++c) {
        if (row_hit >= 0) {
          int blane = (int)(d1s[row_hit * NCOL + c] + 0.5f) - 1;
          int breg = (int)(d2s[row_hit * NCOL + c] + 0.5f) - 1;
          out[tag * (1 + NCOL) + 1 + c] = blane * 16 + breg;
        } else {
          out[tag * (1 + NCOL) + 1 + c] = -1;
        }
      }
    }
    __syncthreads();
  }
}

int main() {
  {
    int* d;
    hipMalloc(&d, 512 * 33 * sizeof(int));
    hipLaunchKernelGGL((probe<32, 32>), dim3(1), dim3(64), 0, 0, d);
    std::vector<int> h(512 * 33);
    hipMemcpy(h.data(), d, h.size() * sizeof(int), hipMemcpyDeviceToHost);
    printf("SHAPE 32x32x16\n");
    for (int t = 0; t < 512; ++t) {
      printf("%d,%d", t, h[t * 33]);
      for (int c = 0; c < 32; ++c) printf(",%d", h[t * 33 + 1 + c]);
      printf("\n");
    }
    hipFree(d);
  }
  {
    int* d;
    hipMalloc(&d, 512 * 17 * sizeof(int));
    hipLaunchKernelGGL((probe<16, 16>), dim3(1), dim3(64), 0, 0, d);
    std::vector<int> h(512 * 17);
    hipMemcpy(h.data(), d, h.size() * sizeof(int), hipMemcpyDeviceToHost);
    printf("SHAPE 16x16x32\n");
    for (int t = 0; t < 512; ++t) {
      printf("%d,%d", t, h[t * 17]);
      for (int c = 0; c < 16; ++c) printf(",%d", h[t * 17 + 1 + c]);
      printf("\n");
    }
    hipFree(d);
  }
  printf("done\n");
  return 0;
}
