// Standalone ablation probe for the flash-bwd dkdv kernel (no torch).
//   hipcc --offload-arch=gfx950 -O3 scripts/bwd_ablate.hip -o /tmp/bwdab
#include "../deepspeed_amd/ops/csrc/attention_bwd.hip"
#include <cstdio>
#include <vector>

__global__ void fill_rand(short* p, long long n, unsigned seed) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    unsigned x = (unsigned)(i * 2654435761u) ^ seed;
    x ^= x >> 13; x *= 0x5bd1e995u; x ^= x >> 15;
    float v = ((float)(x & 0xffff) / 32768.f) - 1.f;   // [-1, 1)
    union { float f; unsigned u; } c; c.f = v;
    p[i] = (short)(c.u >> 16);
  }
}
__global__ void fill_f32(float* p, long long n, float val) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = val;
}

int main() {
  const int B = 8, S = 4096, H = 32, Hkv = 8, Dh = 128;
  const float scale = 0.0883883f;
  size_t nq = (size_t)B * H * S * Dh, nk = (size_t)B * Hkv * S * Dh;
  short *q, *k_, *v, *dout, *qt, *dot, *dk, *dv;
  float *lse, *delta;
  hipMalloc(&q, nq * 2); hipMalloc(&dout, nq * 2); hipMalloc(&qt, nq * 2);
  hipMalloc(&dot, nq * 2);
  hipMalloc(&k_, nk * 2); hipMalloc(&v, nk * 2);
  hipMalloc(&dk, nk * 2); hipMalloc(&dv, nk * 2);
  hipMalloc(&lse, (size_t)B * H * S * 4); hipMalloc(&delta, (size_t)B * H * S * 4);
  hipMemset(q, 0x3c, nq * 2); hipMemset(k_, 0x3c, nk * 2);
  hipMemset(v, 0x3c, nk * 2); hipMemset(dout, 0x3c, nq * 2);
  hipMemset(qt, 0x3c, nq * 2); hipMemset(dot, 0x3c, nq * 2);
  hipMemset(lse, 0, (size_t)B * H * S * 4);
  hipMemset(delta, 0, (size_t)B * H * S * 4);
  for (int mode = 0; mode < 2; ++mode) {
   if (mode == 1) {  // random data + realistic lse
    fill_rand<<<1024, 256>>>(q, nq, 1); fill_rand<<<1024, 256>>>(dout, nq, 2);
    fill_rand<<<1024, 256>>>(qt, nq, 3); fill_rand<<<1024, 256>>>(dot, nq, 4);
    fill_rand<<<1024, 256>>>(k_, nk, 5); fill_rand<<<1024, 256>>>(v, nk, 6);
    fill_f32<<<1024, 256>>>(lse, (long long)B * H * S, 5.f);
    fill_f32<<<1024, 256>>>(delta, (long long)B * H * S, 0.1f);
    hipDeviceSynchronize();
    printf("-- random data --\n");
   } else printf("-- constant data --\n");
  for (int var : {0, 1, 2, 4, 5, 0}) {
    // warmup
    ds_flash_bwd_dkdv_dbg(q, k_, v, dout, qt, dot, lse, delta, dk, dv,
                          B, S, H, Hkv, scale, var, nullptr);
    hipDeviceSynchronize();
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    for (int i = 0; i < 3; ++i)
      ds_flash_bwd_dkdv_dbg(q, k_, v, dout, qt, dot, lse, delta, dk, dv,
                            B, S, H, Hkv, scale, var, nullptr);
    hipEventRecord(e1);
    hipEventSynchronize(e1);
    float ms; hipEventElapsedTime(&ms, e0, e1);
    printf("var=%d  %7.2f ms\n", var, ms / 3);
  }
  }
  printf("err=%s\n", hipGetErrorString(hipGetLastError()));
  return 0;
}
