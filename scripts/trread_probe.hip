// Probe: empirical lane->element mapping of ds_read_b64_tr_b16 on gfx950,
// and permlane32_swap semantics. Pattern: fill LDS shorts with their own
// index, read via the transpose instruction at candidate address patterns,
// dump what each lane received. Build:
//   hipcc --offload-arch=gfx950 -O2 scripts/trread_probe.hip -o /tmp/trprobe
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(2))) unsigned u32x2;

// One wave. LDS filled with lds_short[i] = i. Each lane issues
// ds_read_b64_tr_b16 at byte address = pat(lane, j) for j=0..NJ-1 and
// writes the 4 received shorts to out[lane][j][0..3].
template <int PAT>
__global__ void trread_probe(short* out, int nj) {
  __shared__ short lds[1024];
  const int lane = threadIdx.x;
  for (int i = lane; i < 1024; i += 64) lds[i] = (short)i;
  __syncthreads();
  for (int j = 0; j < nj; ++j) {
    int elem;
    if (PAT == 0)       elem = lane * 4 + j * 256;                  // linear b64
    else if (PAT == 1)  elem = (lane & 15) + j * 16 + (lane >> 4) * 64;  // m156
    else                elem = ((lane & 15) * 2) + j * 16 + (lane >> 4) * 64;
    unsigned addr = (unsigned)(size_t)(lds + elem);  // low 32b = LDS offset
    u32x2 r;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(r) : "v"(addr));
    short4 s = *(short4*)&r;
    out[(lane * nj + j) * 4 + 0] = s.x;
    out[(lane * nj + j) * 4 + 1] = s.y;
    out[(lane * nj + j) * 4 + 2] = s.z;
    out[(lane * nj + j) * 4 + 3] = s.w;
  }
}

// permlane32_swap probe: lane i holds value i in v0 and 1000+i in v1;
// dump both outputs.
__global__ void permlane_probe(int* out) {
  const int lane = threadIdx.x;
  int v0 = lane, v1 = 1000 + lane;
  // builtin returns {new_vdst, new_src} as int2 per docs
  typedef __attribute__((ext_vector_type(2))) int i32x2;
  i32x2 r = __builtin_amdgcn_permlane32_swap(v0, v1, false, false);
  out[lane * 2 + 0] = r[0];
  out[lane * 2 + 1] = r[1];
}

int main() {
  const int NJ = 4;
  short* out;
  hipMalloc(&out, 64 * NJ * 4 * sizeof(short));
  for (int pat = 0; pat < 3; ++pat) {
    if (pat == 0) hipLaunchKernelGGL(trread_probe<0>, 1, 64, 0, 0, out, NJ);
    if (pat == 1) hipLaunchKernelGGL(trread_probe<1>, 1, 64, 0, 0, out, NJ);
    if (pat == 2) hipLaunchKernelGGL(trread_probe<2>, 1, 64, 0, 0, out, NJ);
    short host[64 * NJ * 4];
    hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
    printf("== PAT %d (addr elem: %s)\n", pat,
           pat == 0 ? "lane*4+j*256" : pat == 1 ? "(l&15)+j*16+(l>>4)*64"
                                                : "(l&15)*2+j*16+(l>>4)*64");
    for (int l = 0; l < 64; ++l) {
      printf("lane %2d:", l);
      for (int j = 0; j < NJ; ++j)
        printf("  j%d[%4d %4d %4d %4d]", j, host[(l * NJ + j) * 4],
               host[(l * NJ + j) * 4 + 1], host[(l * NJ + j) * 4 + 2],
               host[(l * NJ + j) * 4 + 3]);
      printf("\n");
    }
  }
  int* iout;
  hipMalloc(&iout, 64 * 2 * sizeof(int));
  hipLaunchKernelGGL(permlane_probe, 1, 64, 0, 0, iout);
  int ih[128];
  hipMemcpy(ih, iout, sizeof(ih), hipMemcpyDeviceToHost);
  printf("== permlane32_swap(v0=lane, v1=1000+lane, false, false)\n");
  for (int l = 0; l < 64; ++l)
    printf("lane %2d: r0=%4d r1=%4d\n", l, ih[l * 2], ih[l * 2 + 1]);
  hipError_t e = hipGetLastError();
  printf("err=%s\n", hipGetErrorString(e));
  return 0;
}
