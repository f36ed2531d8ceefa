// Flash-attention backward for MI355X (gfx950 / CDNA4) — hand-written MFMA.
//
// Implements docs/flash_bwd_design.md, tile-for-tile identical to the
// CPU-validated blueprint ops/flash_bwd_ref.py (which matches autograd).
// Status: compiled + index-math simulated; GPU numerics validation is the
// first round-2 task — reachable only through the _dbg bindings until then.
//
// Layouts (ALL [B, H(kv), S, D] = BHSD, contiguous):
//   q, o, do: [B, H, S, D]      k, v: [B, Hkv, S, D]
//   qt, kt, dot: transposed copies [B, H(kv), D, S] (wrapper-made) so the
//   dK/dV/dQ MFMAs' B-operands (k-dim = q or kv) read contiguously.
//   lse, delta: [B, H, S] fp32.   dq: [B,H,S,D] fp32? -> bf16 out.
//   dk, dv: [B, Hkv, S, D] bf16 (query-head group summed in registers).
//
// MFMA fragment maps (probe-verified, scripts/mfma_probe.hip):
//   A: row=lane&31, k=reg+8*(lane>>5) · B: col=lane&31, same k
//   C/D: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
//
// Grid: kernel1 (dk/dv): (S/32, Hkv, B), 64 threads (one wave per kv tile;
// the query-head group G=H/Hkv is an inner loop so dk/dv accumulate in
// registers with no atomics). kernel2 (dq): (S/32, H, B).
//
// The C/D -> A-operand change of axis for P^T / dS^T goes through a
// 32x32 bf16 LDS tile: store via the C/D map, read back as A fragments
// (row=lane&31, 8 consecutive k) — 16-byte aligned ds reads.

#include "ds_kernels.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8s;
typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int T = 32;   // tile rows
constexpr int D = 128;  // head dim

__device__ __forceinline__ short f2bf(float f) {
  union {
    float f;
    unsigned u;
  } c;
  c.f = f;
  unsigned r = c.u + 0x7FFF + ((c.u >> 16) & 1);
  return (short)(r >> 16);
}

// C/D register r of a 32x32 tile -> row index
__device__ __forceinline__ int cd_row(int r, int half) {
  return (r & 3) + 8 * (r >> 2) + 4 * half;
}

// store a 32x32 fp32 tile (held as C/D fragments, val[16]) into LDS bf16
// [row][col] row-major; then A-frags read ldsP[lane&31][kk*16+8*half .. +8]
__device__ __forceinline__ void cd_to_lds(const float* val, short* lds,
                                          int col, int half) {
#pragma unroll
  for (int r = 0; r < 16; ++r) lds[cd_row(r, half) * T + col] = f2bf(val[r]);
}

template <bool CAUSAL>
__global__ __launch_bounds__(64) void flash_bwd_dkdv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const short* __restrict__ qt, const short* __restrict__ dot,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv,
    const int B, const int S, const int H, const int Hkv,
    const float scale) {
  __shared__ __align__(16) short ldsP[T * T];
  __shared__ __align__(16) short ldsD[T * T];

  const int lane = threadIdx.x;
  const int col = lane & 31;
  const int half = lane >> 5;
  const int kv0 = blockIdx.x * T;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int G = H / Hkv;
  const int kvrow = kv0 + col;

  // ---- preload K and V A-fragments for this kv tile (row = kvrow)
  bf16x8s kf[8], vf[8];
  {
    const long long base =
        (((long long)b * Hkv + hkv) * S + kvrow) * D + 8 * half;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      kf[kk] = *(const bf16x8s*)(k + base + kk * 16);
      vf[kk] = *(const bf16x8s*)(v + base + kk * 16);
    }
  }

  f32x16 dvacc[4] = {};  // [kv rows x d cols], col=lane&31 = d_local
  f32x16 dkacc[4] = {};

  for (int g = 0; g < G; ++g) {
    const int h = hkv * G + g;
    const long long qbase = (((long long)b * H + h) * S) * D;
    const long long tbase = (((long long)b * H + h) * D) * S;
    const long long sbase = ((long long)b * H + h) * S;

    const int q_start = CAUSAL ? kv0 : 0;
    for (int qs = q_start; qs < S; qs += T) {
      const int qrow = qs + col;
      // B-frags of Q and dO for this q tile (col = qrow, k slots along d)
      bf16x8s qf[8], dof[8];
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        qf[kk] = *(const bf16x8s*)(q + qbase + (long long)qrow * D +
                                   kk * 16 + 8 * half);
        dof[kk] = *(const bf16x8s*)(dout + qbase + (long long)qrow * D +
                                    kk * 16 + 8 * half);
      }

      // S^T = K Q^T ; dP^T = V dO^T
      f32x16 st = {}, dpt = {};
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf[kk], qf[kk], st,
                                                     0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf[kk], dof[kk], dpt,
                                                      0, 0, 0);
      }

      const float l = lse[sbase + qrow];
      const float dl = delta[sbase + qrow];
      float pt[16], dst[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvl = cd_row(r, half);
        float s = st[r] * scale;
        const bool dead = CAUSAL && (kv0 + kvl > qrow);
        pt[r] = dead ? 0.f : __expf(s - l);
        dst[r] = pt[r] * (dpt[r] - dl) * scale;
      }

      // transpose P^T and dS^T through LDS into A-operand layout
      __syncthreads();
      cd_to_lds(pt, ldsP, col, half);
      cd_to_lds(dst, ldsD, col, half);
      __syncthreads();

      // dV[kv][d] += P^T(k=q) dO^T-asB ; dK[kv][d] += dS^T(k=q) Q^T-asB
#pragma unroll
      for (int dblk = 0; dblk < 4; ++dblk) {
        const long long trow = (long long)(dblk * 32 + col) * S + qs;
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          bf16x8s pA = *(const bf16x8s*)(ldsP + (lane & 31) * T + kk * 16 +
                                         8 * half);
          bf16x8s dA = *(const bf16x8s*)(ldsD + (lane & 31) * T + kk * 16 +
                                         8 * half);
          bf16x8s doB = *(const bf16x8s*)(dot + tbase + trow + kk * 16 +
                                          8 * half);
          bf16x8s qB = *(const bf16x8s*)(qt + tbase + trow + kk * 16 +
                                         8 * half);
          dvacc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pA, doB, dvacc[dblk], 0, 0, 0);
          dkacc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              dA, qB, dkacc[dblk], 0, 0, 0);
        }
      }
    }
  }

  // ---- write dk/dv (C/D: col = d_local, row = kv via reg map)
  const long long obase = (((long long)b * Hkv + hkv) * S) * D;
#pragma unroll
  for (int dblk = 0; dblk < 4; ++dblk) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kvl = cd_row(r, half);
      const long long off = obase + (long long)(kv0 + kvl) * D +
                            dblk * 32 + col;
      dv[off] = f2bf(dvacc[dblk][r]);
      dk[off] = f2bf(dkacc[dblk][r]);
    }
  }
}

template <bool CAUSAL>
__global__ __launch_bounds__(64) void flash_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const short* __restrict__ kt,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq,
    const int B, const int S, const int H, const int Hkv,
    const float scale) {
  __shared__ __align__(16) short ldsD[T * T];

  const int lane = threadIdx.x;
  const int col = lane & 31;
  const int half = lane >> 5;
  const int qs = blockIdx.x * T;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (H / Hkv);
  const int qrow = qs + col;

  const long long qbase = (((long long)b * H + h) * S) * D;
  const long long kbase = (((long long)b * Hkv + hkv) * S) * D;
  const long long ktbase = (((long long)b * Hkv + hkv) * D) * S;
  const long long sbase = ((long long)b * H + h) * S;

  // B-frags of Q and dO for my q tile
  bf16x8s qf[8], dof[8];
#pragma unroll
  for (int kk = 0; kk < 8; ++kk) {
    qf[kk] = *(const bf16x8s*)(q + qbase + (long long)qrow * D + kk * 16 +
                               8 * half);
    dof[kk] = *(const bf16x8s*)(dout + qbase + (long long)qrow * D +
                                kk * 16 + 8 * half);
  }
  const float l = lse[sbase + qrow];
  const float dl = delta[sbase + qrow];

  f32x16 dqacc[4] = {};  // rows=q, cols=d (col=lane&31=d_local)

  const int kv_end = CAUSAL ? min(S, qs + T) : S;
  for (int kv0 = 0; kv0 < kv_end; kv0 += T) {
    // A-frags of K and V for this kv tile (row = kv0+col)
    bf16x8s kf[8], vf[8];
    const long long base = kbase + (long long)(kv0 + col) * D + 8 * half;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      kf[kk] = *(const bf16x8s*)(k + base + kk * 16);
      vf[kk] = *(const bf16x8s*)(v + base + kk * 16);
    }
    f32x16 st = {}, dpt = {};
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf[kk], qf[kk], st,
                                                   0, 0, 0);
      dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf[kk], dof[kk], dpt,
                                                    0, 0, 0);
    }
    float dst[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kvl = cd_row(r, half);
      float s = st[r] * scale;
      const bool dead = CAUSAL && (kv0 + kvl > qrow);
      const float pt = dead ? 0.f : __expf(s - l);
      dst[r] = pt * (dpt[r] - dl) * scale;
    }

    // dS^T (rows kv, cols q) -> LDS -> read as dS A-frags (rows q, k=kv):
    // store transposed: lds[q][kv] = dst  (q = col, kv = cd_row)
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 16; ++r)
      ldsD[col * T + cd_row(r, half)] = f2bf(dst[r]);
    __syncthreads();

    // dQ[q][d] += dS(row=q, k=kv) · K^T-asB(k=kv, col=d)
#pragma unroll
    for (int dblk = 0; dblk < 4; ++dblk) {
      const long long trow = ktbase + (long long)(dblk * 32 + col) * S + kv0;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8s dA = *(const bf16x8s*)(ldsD + (lane & 31) * T + kk * 16 +
                                       8 * half);
        bf16x8s kB = *(const bf16x8s*)(kt + trow + kk * 16 + 8 * half);
        dqacc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            dA, kB, dqacc[dblk], 0, 0, 0);
      }
    }
  }

  // write dq (rows=q via reg map, col=d_local)
#pragma unroll
  for (int dblk = 0; dblk < 4; ++dblk) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int ql = cd_row(r, half);
      dq[qbase + (long long)(qs + ql) * D + dblk * 32 + col] =
          f2bf(dqacc[dblk][r]);
    }
  }
}

}  // namespace

extern "C" void ds_flash_bwd(const void* q, const void* k, const void* v,
                             const void* dout, const void* qt,
                             const void* kt, const void* dot,
                             const float* lse, const float* delta, void* dq,
                             void* dk, void* dv, int B, int S, int H,
                             int Hkv, float scale, int causal, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  dim3 g1((S + T - 1) / T, Hkv, B);
  dim3 g2((S + T - 1) / T, H, B);
  if (causal) {
    hipLaunchKernelGGL((flash_bwd_dkdv_kernel<true>), g1, dim3(64), 0, st,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const short*)qt,
                       (const short*)dot, lse, delta, (short*)dk, (short*)dv,
                       B, S, H, Hkv, scale);
    hipLaunchKernelGGL((flash_bwd_dq_kernel<true>), g2, dim3(64), 0, st,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const short*)kt, lse, delta,
                       (short*)dq, B, S, H, Hkv, scale);
  } else {
    hipLaunchKernelGGL((flash_bwd_dkdv_kernel<false>), g1, dim3(64), 0, st,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const short*)qt,
                       (const short*)dot, lse, delta, (short*)dk, (short*)dv,
                       B, S, H, Hkv, scale);
    hipLaunchKernelGGL((flash_bwd_dq_kernel<false>), g2, dim3(64), 0, st,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const short*)kt, lse, delta,
                       (short*)dq, B, S, H, Hkv, scale);
  }
}
