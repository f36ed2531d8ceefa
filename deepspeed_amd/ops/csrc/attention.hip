// Flash-attention forward for MI355X (gfx950 / CDNA4) — hand-written MFMA.
//
// Replaces the aotriton SDPA forward on the inference/prefill path
// (reference analogue: csrc/transformer/inference softmax+GEMM chain and
// inference/v2 blocked flash). Structure follows the CDNA4 guide's
// verified flash ladder (cdna_hip_programming.md §attention): swapped
// QK^T so softmax is per-lane, online-softmax rescale, MFMA everywhere.
//
// Design (v1, correctness-first):
// * one WAVE per 32 query rows; grid (ceil(S/32), H, B), 64 threads.
//   No LDS at all — K is read as A-fragments straight from global (the
//   guide's pitfall #7: at these sizes L2 serves K/V better than staging),
//   V is consumed from a pre-transposed copy vt[B,Hkv,D,S] so its
//   A-fragments are contiguous 16-byte loads.
// * mfma_f32_32x32x16_bf16 with probe-verified layouts
//   (scripts/mfma_probe.hip, run on gfx950 2026-08-20):
//     A: row = lane&31, k = reg + 8*(lane>>5)
//     B: col = lane&31, k = reg + 8*(lane>>5)
//     C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// * swapped S^T = mfma(A=K, B=Q): each lane then owns 16 score entries of
//   ONE query column -> row max/sum needs 15 VALU ops + one shfl_xor(32).
// * P re-layout for PV (C/D -> B fragment) is a register permutation
//   within the (lane, lane^32) pair: reg' = (kv&3) + 4*(kv>>3),
//   cross-half values fetched with shfl_xor(32).
//
// Layouts: q,k [B,S,H(kv),128] bf16; vt [B,Hkv,128,S] bf16; o [B,S,H,128].
// Requires D=128, S % 32 == 0 (wrapper falls back to SDPA otherwise).

#include "ds_kernels.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8s;
typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int QB = 32;   // query rows per wave
constexpr int KB = 32;   // kv rows per tile
constexpr int D = 128;   // head dim

__device__ __forceinline__ float bf2f(short s) {
  union {
    float f;
    unsigned u;
  } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return c.f;
}

__device__ __forceinline__ short f2bf(float f) {
  union {
    float f;
    unsigned u;
  } c;
  c.f = f;
  unsigned r = c.u + 0x7FFF + ((c.u >> 16) & 1);
  return (short)(r >> 16);
}

template <bool CAUSAL>
__global__ __launch_bounds__(64) void flash_fwd_kernel(
    const short* __restrict__ q,   // [B, S, H, D]
    const short* __restrict__ k,   // [B, S, Hkv, D]
    const short* __restrict__ vt,  // [B, Hkv, D, S]
    short* __restrict__ o,         // [B, S, H, D]
    const int B, const int S, const int H, const int Hkv,
    const float scale) {
  const int lane = threadIdx.x;
  const int q0 = blockIdx.x * QB;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (H / Hkv);
  if (q0 >= S) return;

  const int col = lane & 31;        // my query row within the tile
  const int half = lane >> 5;       // k-slot half
  const int qrow = q0 + col;

  // ---- preload Q as B-fragments: qf[kk] holds d = kk*16 + 8*half + [0,8)
  bf16x8s qf[8];
  {
    const short* qp = q + (((long long)b * S + qrow) * H + h) * D;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      qf[kk] = *(const bf16x8s*)(qp + kk * 16 + 8 * half);
  }

  float m_run = -1e30f;
  float l_run = 0.f;
  f32x16 oacc[4] = {};  // O^T accumulators, one per 32-wide d block

  const int kv_end = CAUSAL ? min(S, q0 + QB) : S;
  const long long k_bh = ((long long)b * S) * Hkv + hkv;  // row stride below
  const short* vtp = vt + (((long long)b * Hkv + hkv) * D) * S;

  for (int kv0 = 0; kv0 < kv_end; kv0 += KB) {
    // ---- S^T = K · Q^T : A = K rows (kv), B = Q cols (q)
    f32x16 st = {};
    const short* kp = k + ((k_bh + (long long)kv0 * Hkv) * D);
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      bf16x8s kf = *(const bf16x8s*)(kp + (long long)col * Hkv * D +
                                     kk * 16 + 8 * half);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[kk], st, 0, 0, 0);
    }

    // ---- masked scale + online softmax (per-lane: one q, 16 kv entries)
    float p[16];
    float mt = -1e30f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kvl = (r & 3) + 8 * (r >> 2) + 4 * half;
      float s = st[r] * scale;
      if (CAUSAL && kv0 + kvl > qrow) s = -1e30f;
      p[r] = s;
      mt = fmaxf(mt, s);
    }
    mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
    const float m_new = fmaxf(m_run, mt);
    const float alpha = __expf(m_run - m_new);
    float lt = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      p[r] = __expf(p[r] - m_new);
      lt += p[r];
    }
    lt += __shfl_xor(lt, 32, 64);
    l_run = l_run * alpha + lt;
    m_run = m_new;

    // other half's p values (same q column lives in lane^32)
    float px[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) px[r] = __shfl_xor(p[r], 32, 64);

    // ---- build P B-fragments: slot (kk2, reg) needs kv = kk2*16+8*half+reg
    bf16x8s pf[2];
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
#pragma unroll
      for (int reg = 0; reg < 8; ++reg) {
        const int kv = kk2 * 16 + 8 * half + reg;
        const int rp = (kv & 3) + 4 * (kv >> 3);
        const bool mine = (((kv >> 2) & 1) == half);
        pf[kk2][reg] = f2bf(mine ? p[rp] : px[rp]);
      }
    }

    // ---- O^T += V^T · P : A = V^T rows (d), k = kv; B = P cols (q)
#pragma unroll
    for (int dblk = 0; dblk < 4; ++dblk) {
      const short* vp = vtp + (long long)(dblk * 32 + col) * S + kv0;
      // rescale accumulated output by alpha (once per kv tile)
#pragma unroll
      for (int r = 0; r < 16; ++r) oacc[dblk][r] *= alpha;
#pragma unroll
      for (int kk2 = 0; kk2 < 2; ++kk2) {
        bf16x8s vf = *(const bf16x8s*)(vp + kk2 * 16 + 8 * half);
        oacc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            vf, pf[kk2], oacc[dblk], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O[q][d] = O^T / l
  const float inv_l = 1.f / l_run;
  short* op = o + (((long long)b * S + qrow) * H + h) * D;
#pragma unroll
  for (int dblk = 0; dblk < 4; ++dblk) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int d = dblk * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
      op[d] = f2bf(oacc[dblk][r] * inv_l);
    }
  }
}

}  // namespace

extern "C" void ds_flash_fwd(const void* q, const void* k, const void* vt,
                             void* o, int B, int S, int H, int Hkv,
                             float scale, int causal, void* stream) {
  dim3 grid((S + QB - 1) / QB, H, B);
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (causal)
    hipLaunchKernelGGL((flash_fwd_kernel<true>), grid, dim3(64), 0, st,
                       (const short*)q, (const short*)k, (const short*)vt,
                       (short*)o, B, S, H, Hkv, scale);
  else
    hipLaunchKernelGGL((flash_fwd_kernel<false>), grid, dim3(64), 0, st,
                       (const short*)q, (const short*)k, (const short*)vt,
                       (short*)o, B, S, H, Hkv, scale);
}
