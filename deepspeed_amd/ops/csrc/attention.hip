// Flash-attention forward for MI355X (gfx950 / CDNA4) — hand-written MFMA.
//
// Replaces the aotriton SDPA forward on the inference/prefill path
// (reference analogue: csrc/transformer/inference softmax+GEMM chain and
// inference/v2 blocked flash). Structure follows the CDNA4 guide's
// verified flash ladder (cdna_hip_programming.md §attention).
//
// v2 structure (8-wave K/V reuse):
// * 512-thread workgroup = 8 waves x 32 query rows (Q tile = 256 rows);
//   grid (ceil(S/256), H, B). K and V^T tiles are staged in LDS ONCE per
//   kv step and consumed by all 8 waves — global K/V traffic drops 8x vs
//   the naive per-wave version (which measured HBM-bound at 164 TF).
// * XOR swizzle on the LDS tiles (guide §6 G4: `byte ^= ((row&7)<<4)`)
//   so the MFMA fragment reads (row-strided ds_read_b128) don't bank
//   conflict.
// * mfma_f32_32x32x16_bf16 with probe-verified layouts
//   (scripts/mfma_probe.hip, run on gfx950 2026-08-20):
//     A: row = lane&31, k = reg + 8*(lane>>5)
//     B: col = lane&31, k = reg + 8*(lane>>5)
//     C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// * swapped S^T = mfma(A=K, B=Q): each lane owns 16 score entries of ONE
//   query column -> row max/sum is 15 VALU ops + one shfl_xor(32).
// * P re-layout for PV (C/D -> B fragment) is a register permutation
//   within the (lane, lane^32) pair: reg' = (kv&3) + 4*(kv>>3),
//   cross-half values via shfl_xor(32).
// * KSUB template: number of 32-row kv sub-tiles staged per barrier pair
//   (KSUB=2 = KVBLK 64, half the barriers — guide ladder step; dbg-only
//   until GPU-validated, default stays the validated KSUB=1).
//
// Layouts: q,k [B,S,H(kv),128] bf16; vt [B,Hkv,128,S] bf16; o [B,S,H,128].
// Requires D=128, S % 32 == 0 (wrapper falls back to SDPA otherwise).

#include "ds_kernels.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8s;
typedef __attribute__((ext_vector_type(8))) short lds_chunk;  // 16 B
typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int QB = 32;    // query rows per wave
constexpr int NWAVE = 8;  // waves per workgroup
constexpr int QTILE = QB * NWAVE;
constexpr int TPB = NWAVE * 64;  // 8 waves x 64 lanes
constexpr int D = 128;           // head dim

__device__ __forceinline__ short f2bf(float f) {
  union {
    float f;
    unsigned u;
  } c;
  c.f = f;
  unsigned r = c.u + 0x7FFF + ((c.u >> 16) & 1);
  return (short)(r >> 16);
}

// LDS tile helpers: tiles stored as 16-byte chunks with an XOR swizzle on
// the chunk index so row-strided fragment reads spread across banks.
// K tile: [KSUB*32][D] bf16 = rows x 16 chunks. V^T: [D][KSUB*32] = 128 x
// KSUB*4 chunks per row.
__device__ __forceinline__ int k_sw(int row, int chunk) {
  return row * 16 + (chunk ^ (row & 7));
}
template <int W>
__device__ __forceinline__ int v_sw(int row, int chunk) {
  return row * W + (chunk ^ (row & (W - 1)));
}

// VAR ablation (debug): bit0 = K from LDS, bit1 = V from LDS (3 = normal)
template <bool CAUSAL, int VAR = 3, int KSUB = 1>
__global__ __launch_bounds__(TPB) void flash_fwd_kernel(
    const short* __restrict__ q,   // [B, S, H, D]
    const short* __restrict__ k,   // [B, S, Hkv, D]
    const short* __restrict__ vt,  // [B, Hkv, D, S]
    short* __restrict__ o,         // [B, S, H, D]
    float* __restrict__ lse,       // optional [B, H, S] log-sum-exp
    const int B, const int S, const int H, const int Hkv,
    const float scale) {
  constexpr int KB = KSUB * 32;   // kv rows staged per barrier pair
  constexpr int VW = KSUB * 4;    // V^T chunks per row
  __shared__ lds_chunk kt_lds[KB * 16];
  __shared__ lds_chunk vt_lds[D * VW];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int q0b = blockIdx.x * QTILE;  // block's first q row
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (H / Hkv);

  const int col = lane & 31;  // my query row within the wave / kv row in A
  const int half = lane >> 5;
  const int q0w = q0b + wid * QB;
  const int qrow = q0w + col;
  const bool q_ok = qrow < S;
  const int qload = q_ok ? qrow : S - 1;

  // ---- preload Q as B-fragments: qf[kk] holds d = kk*16 + 8*half + [0,8)
  bf16x8s qf[8];
  {
    const short* qp = q + (((long long)b * S + qload) * H + h) * D;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      qf[kk] = *(const bf16x8s*)(qp + kk * 16 + 8 * half);
  }

  float m_run = -1e30f;
  float l_run = 0.f;
  f32x16 oacc[4] = {};  // O^T accumulators, one per 32-wide d block

  const int kv_end_blk = CAUSAL ? min(S, q0b + QTILE) : S;
  const int kv_end_wave = CAUSAL ? min(S, q0w + QB) : S;
  const long long k_base = (((long long)b * S) * Hkv + hkv) * D;
  const short* vtp = vt + (((long long)b * Hkv + hkv) * D) * S;

  for (int kv0 = 0; kv0 < kv_end_blk; kv0 += KB) {
    // ---- cooperative stage: K tile (KB*16 chunks) + V^T (D*VW chunks)
    __syncthreads();
    for (int g = tid; g < KB * 16; g += TPB) {
      const int row = g >> 4, c = g & 15;
      const int kvr = min(kv0 + row, S - 1);
      kt_lds[k_sw(row, c)] = *(const bf16x8s*)(
          k + k_base + (long long)kvr * Hkv * D + c * 8);
    }
    for (int g = tid; g < D * VW; g += TPB) {
      const int vrow = g / VW, vc = g % VW;
      const int kvc = min(kv0 + vc * 8, S - 8);  // S%8==0 guaranteed
      vt_lds[v_sw<VW>(vrow, vc)] = *(const bf16x8s*)(
          vtp + (long long)vrow * S + kvc);
    }
    __syncthreads();

    if (kv0 >= kv_end_wave) continue;  // past my diagonal: barriers only

#pragma unroll
    for (int sub = 0; sub < KSUB; ++sub) {
      const int kv0s = kv0 + sub * 32;
      if (KSUB > 1 && kv0s >= kv_end_wave) break;  // no barriers inside

      // ---- S^T = K · Q^T : A = K rows (kv), B = Q cols (q)
      f32x16 st = {};
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        // A slot: row = sub*32+col, d chunk = kk*2 + half (8 bf16 each)
        bf16x8s kf;
        if (VAR & 1)
          kf = kt_lds[k_sw(sub * 32 + col, kk * 2 + half)];
        else
          kf = *(const bf16x8s*)(k + k_base +
                                 (long long)min(kv0s + col, S - 1) * Hkv * D +
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
        if ((CAUSAL && kv0s + kvl > qrow) || kv0s + kvl >= S) s = -1e30f;
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

      // ---- build P B-fragments: slot (kk2, reg) -> kv = kk2*16+8*half+reg
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
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[dblk][r] *= alpha;
#pragma unroll
        for (int kk2 = 0; kk2 < 2; ++kk2) {
          // A slot: row = dblk*32+col, kv chunk = sub*4 + kk2*2 + half
          bf16x8s vf;
          if (VAR & 2)
            vf = vt_lds[v_sw<VW>(dblk * 32 + col, sub * 4 + kk2 * 2 + half)];
          else
            vf = *(const bf16x8s*)(vtp + (long long)(dblk * 32 + col) * S +
                                   kv0s + kk2 * 16 + 8 * half);
          oacc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              vf, pf[kk2], oacc[dblk], 0, 0, 0);
        }
      }
    }
  }

  // ---- epilogue: O[q][d] = O^T / l
  if (!q_ok) return;
  const float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
  if (lse != nullptr && half == 0)  // one lane half owns the q row
    lse[((long long)b * H + h) * S + qrow] = m_run + __logf(l_run);
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
                             void* o, void* lse, int B, int S, int H,
                             int Hkv, float scale, int causal, void* stream) {
  dim3 grid((S + QTILE - 1) / QTILE, H, B);
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (causal)
    hipLaunchKernelGGL((flash_fwd_kernel<true>), grid, dim3(TPB), 0, st,
                       (const short*)q, (const short*)k, (const short*)vt,
                       (short*)o, (float*)lse, B, S, H, Hkv, scale);
  else
    hipLaunchKernelGGL((flash_fwd_kernel<false>), grid, dim3(TPB), 0, st,
                       (const short*)q, (const short*)k, (const short*)vt,
                       (short*)o, (float*)lse, B, S, H, Hkv, scale);
}

// variants 0-3: K/V LDS ablation at KSUB=1; variant 4: KVBLK=64 (KSUB=2)
extern "C" void ds_flash_fwd_dbg(const void* q, const void* k, const void* vt,
                                 void* o, int B, int S, int H, int Hkv,
                                 float scale, int variant, void* stream) {
  dim3 grid((S + QTILE - 1) / QTILE, H, B);
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  switch (variant) {
    case 0:
      hipLaunchKernelGGL((flash_fwd_kernel<true, 0>), grid, dim3(TPB), 0, st,
                         (const short*)q, (const short*)k, (const short*)vt,
                         (short*)o, nullptr, B, S, H, Hkv, scale);
      break;
    case 1:
      hipLaunchKernelGGL((flash_fwd_kernel<true, 1>), grid, dim3(TPB), 0, st,
                         (const short*)q, (const short*)k, (const short*)vt,
                         (short*)o, nullptr, B, S, H, Hkv, scale);
      break;
    case 2:
      hipLaunchKernelGGL((flash_fwd_kernel<true, 2>), grid, dim3(TPB), 0, st,
                         (const short*)q, (const short*)k, (const short*)vt,
                         (short*)o, nullptr, B, S, H, Hkv, scale);
      break;
    case 4:
      hipLaunchKernelGGL((flash_fwd_kernel<true, 3, 2>), grid, dim3(TPB), 0,
                         st, (const short*)q, (const short*)k,
                         (const short*)vt, (short*)o, nullptr, B, S, H, Hkv,
                         scale);
      break;
    default:
      hipLaunchKernelGGL((flash_fwd_kernel<true, 3>), grid, dim3(TPB), 0, st,
                         (const short*)q, (const short*)k, (const short*)vt,
                         (short*)o, nullptr, B, S, H, Hkv, scale);
  }
}
