// Flash-attention forward for MI355X (gfx950 / CDNA4) — hand-written MFMA.
//
// Replaces the aotriton SDPA forward on the training/prefill hot path
// (reference analogue: csrc/transformer/inference softmax+GEMM chain and
// inference/v2 blocked flash). v3 follows the CDNA4 guide's verified
// 8-wave/QB32/KVBLK64 attention ladder (cdna_hip_programming.md §B).
//
// v3 structure (fixes v2's 144 B/lane scratch spill and 2-barrier stalls):
// * 512-thread workgroup = 8 waves x 32 query rows (Q tile = 256 rows);
//   grid (ceil(S/256), H, B). KVBLK = 64 rows per step.
// * K and V^T tiles double-buffered in LDS (64 KB total): ONE
//   __syncthreads per 64-row kv step instead of two per 32-row step.
// * async-STAGE split: next tile's global loads are issued into registers
//   BEFORE the current tile's compute (HBM latency hides under MFMA),
//   ds_writes land after the compute, before the barrier.
// * XOR swizzle on both LDS tiles (`chunk ^= row&7`) — bank-conflict-free
//   row-strided ds_read_b128 fragment reads.
// * mfma_f32_32x32x16_bf16 with probe-verified layouts
//   (scripts/mfma_probe.hip, run on gfx950 2026-08-20):
//     A: row = lane&31, k = reg + 8*(lane>>5)
//     B: col = lane&31, k = reg + 8*(lane>>5)
//     C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// * swapped S^T = mfma(A=K, B=Q): each lane owns 32 score entries of ONE
//   query column -> row max/sum is ~31 VALU ops + one shfl_xor(32).
// * P re-layout for PV (C/D -> B fragment) with COMPILE-TIME register
//   indices (v2 used half-dependent runtime indices -> scratch): for
//   pf[ks][reg], the owner half is ho=(reg>>2)&1 and both candidate
//   source registers rp = (reg&3)+8*(ks&1)+4*{ho,1-ho} are constants;
//   the cross-half value moves with one shfl_xor(32).
// * defer-max (RESCALE_THRESHOLD=8): skip the O(64-VALU) rescale pass
//   when the tile max doesn't raise the running max by >8 (exp stays
//   bounded by e^8, fp32 accumulators don't care).
// * s_setprio(1) around the MFMA clusters.
//
// Layouts: q,k [B,S,H(kv),128] bf16; vt [B,Hkv,128,S] bf16; o [B,S,H,128].
// Requires D=128, S % 32 == 0 (wrapper falls back to SDPA otherwise).

#include "ds_kernels.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8s;
typedef __attribute__((ext_vector_type(8))) short lds_chunk;  // 16 B
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef __attribute__((ext_vector_type(2))) int i32x2;

constexpr int QB = 32;    // query rows per wave
constexpr int NWAVE = 8;  // waves per workgroup
constexpr int QTILE = QB * NWAVE;
constexpr int TPB = NWAVE * 64;  // 8 waves x 64 lanes
constexpr int D = 128;           // head dim
constexpr int KB = 64;           // kv rows per step

__device__ __forceinline__ short f2bf(float f) {
  union {
    float f;
    unsigned u;
  } c;
  c.f = f;
  unsigned r = c.u + 0x7FFF + ((c.u >> 16) & 1);
  return (short)(r >> 16);
}

// Swizzled chunk index inside a tile: row-major rows of NC 16-byte chunks,
// chunk XOR'd with row&7 so row-strided fragment reads spread the banks.
__device__ __forceinline__ int k_sw(int row, int chunk) {
  return row * 16 + (chunk ^ (row & 7));
}
__device__ __forceinline__ int v_sw(int row, int chunk) {
  return row * 8 + (chunk ^ (row & 7));
}

struct StageRegs {
  bf16x8s k0, k1, v0, v1;
};

// Issue the global loads for one kv tile (2 K chunks + 2 V^T chunks per
// thread). Called EARLY so the HBM latency hides under the MFMA phase.
__device__ __forceinline__ StageRegs stage_load(
    const short* __restrict__ k, const short* __restrict__ vtp,
    long long k_base, int HkvD, int kv0, int S, int tid) {
  StageRegs r;
  const int krow = tid >> 4, kc = tid & 15;
  const long long kr0 = min(kv0 + krow, S - 1);
  const long long kr1 = min(kv0 + krow + 32, S - 1);
  r.k0 = *(const bf16x8s*)(k + k_base + kr0 * HkvD + kc * 8);
  r.k1 = *(const bf16x8s*)(k + k_base + kr1 * HkvD + kc * 8);
  const int vrow = tid >> 3, vc = tid & 7;
  const int kvc = min(kv0 + vc * 8, S - 8);
  r.v0 = *(const bf16x8s*)(vtp + (long long)vrow * S + kvc);
  r.v1 = *(const bf16x8s*)(vtp + (long long)(vrow + 64) * S + kvc);
  return r;
}

__device__ __forceinline__ void stage_write(const StageRegs& r,
                                            lds_chunk* kbuf, lds_chunk* vbuf,
                                            int tid) {
  const int krow = tid >> 4, kc = tid & 15;
  kbuf[k_sw(krow, kc)] = r.k0;
  kbuf[k_sw(krow + 32, kc)] = r.k1;
  const int vrow = tid >> 3, vc = tid & 7;
  vbuf[v_sw(vrow, vc)] = r.v0;
  vbuf[v_sw(vrow + 64, vc)] = r.v1;
}

template <bool CAUSAL, bool DEFER = true>
__global__ __launch_bounds__(TPB) void flash_fwd_kernel(
    const short* __restrict__ q,   // [B, S, H, D]
    const short* __restrict__ k,   // [B, S, Hkv, D]
    const short* __restrict__ vt,  // [B, Hkv, D, S]
    short* __restrict__ o,         // [B, S, H, D]
    float* __restrict__ lse,       // optional [B, H, S] log-sum-exp
    const int B, const int S, const int H, const int Hkv,
    const float scale) {
  __shared__ lds_chunk kbuf[2][KB * 16];  // K: [64 rows][16 chunks]
  __shared__ lds_chunk vbuf[2][D * 8];    // V^T: [128 rows][8 chunks]

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
  const int HkvD = Hkv * D;
  const long long k_base = (((long long)b * S) * Hkv + hkv) * D;
  const short* vtp = vt + (((long long)b * Hkv + hkv) * D) * S;
  const int nt = (kv_end_blk + KB - 1) / KB;

  // ---- prologue: stage tile 0 into buffer 0
  {
    StageRegs sr = stage_load(k, vtp, k_base, HkvD, 0, S, tid);
    stage_write(sr, kbuf[0], vbuf[0], tid);
  }
  __syncthreads();

  for (int it = 0; it < nt; ++it) {
    const int kv0 = it * KB;
    const int cur = it & 1;

    // ---- async stage: issue next tile's loads before compute
    StageRegs sr;
    const bool have_next = (it + 1 < nt);
    if (have_next)
      sr = stage_load(k, vtp, k_base, HkvD, kv0 + KB, S, tid);

    if (kv0 < kv_end_wave) {
      const lds_chunk* kt_lds = kbuf[cur];
      const lds_chunk* vt_lds = vbuf[cur];

      // ---- S^T = K · Q^T for both 32-row kv subtiles
      f32x16 p2[2];
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        f32x16 st = {};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          bf16x8s kf = kt_lds[k_sw(sub * 32 + col, kk * 2 + half)];
          st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[kk], st,
                                                       0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
        p2[sub] = st;
      }

      // ---- masked scale + tile max (lane owns 32 kv entries of one q).
      // Interior tiles (fully below the causal diagonal for every q row
      // of this wave, fully inside S) skip the 64 mask compares.
      float mt = -1e30f;
      const bool interior = kv0 + KB <= (CAUSAL ? q0w + 1 : S) &&
                            kv0 + KB <= S;
      if (interior) {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float s = p2[sub][r] * scale;
            p2[sub][r] = s;
            mt = fmaxf(mt, s);
          }
        }
      } else {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kvl = kv0 + sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
            float s = p2[sub][r] * scale;
            if ((CAUSAL && kvl > qrow) || kvl >= S) s = -1e30f;
            p2[sub][r] = s;
            mt = fmaxf(mt, s);
          }
        }
      }
      mt = fmaxf(mt, __shfl_xor(mt, 32, 64));

      // ---- online softmax with defer-max
      bool rescale = true;
      float m_new;
      if (DEFER) {
        rescale = !__all(mt <= m_run + 8.f);
        m_new = rescale ? fmaxf(m_run, mt) : m_run;
      } else {
        m_new = fmaxf(m_run, mt);
      }
      float lt = 0.f;
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          p2[sub][r] = __expf(p2[sub][r] - m_new);
          lt += p2[sub][r];
        }
      }
      lt += __shfl_xor(lt, 32, 64);
      const float alpha = __expf(m_run - m_new);
      l_run = (rescale ? l_run * alpha : l_run) + lt;
      m_run = m_new;

      // ---- build P B-fragments with compile-time register indices.
      // pf[ks] covers kv = ks*16 + reg + 8*half; the owner half of that
      // score is ho=(reg>>2)&1 (independent of the requester's half), at
      // register rp = (reg&3)+8*(ks&1)+4*h of p2[ks>>1]. One
      // permlane32_swap(a=low-half need, b=high-half need) delivers BOTH
      // fragment slots (reg=rr at r0, reg=rr+4 at r1) — 16 VALU-pipe
      // permlanes per tile instead of 32 LDS-pipe ds_bpermutes.
      bf16x8s pf[4];
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          const int s_ = ks >> 1;                   // compile-time
          const int rpA = rr + 8 * (ks & 1);        // h=0 requester's need
          const int rpB = rpA + 4;                  // h=1 requester's need
          // inline asm: the __builtin_amdgcn_permlane32_swap intrinsic is
          // wrongly CSE'd across calls with different args by this LLVM
          // (16 calls collapse to 2 — see /tmp repro pl3/pl6, 2026-08-20)
          float a = p2[s_][rpA], b = p2[s_][rpB];
          asm("v_permlane32_swap_b32 %0, %1" : "+v"(a), "+v"(b));
          pf[ks][rr] = f2bf(a);
          pf[ks][rr + 4] = f2bf(b);
        }
      }

      // ---- O^T += V^T · P : A = V^T rows (d), k = kv; B = P cols (q)
#pragma unroll
      for (int dblk = 0; dblk < 4; ++dblk) {
        if (rescale) {
#pragma unroll
          for (int r = 0; r < 16; ++r) oacc[dblk][r] *= alpha;
        }
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          bf16x8s vf = vt_lds[v_sw(dblk * 32 + col, ks * 2 + half)];
          oacc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              vf, pf[ks], oacc[dblk], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
    }

    // ---- write next tile into the other buffer; barrier flips it live
    if (have_next)
      stage_write(sr, kbuf[cur ^ 1], vbuf[cur ^ 1], tid);
    __syncthreads();
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

// dbg variants: 0 = defer-max OFF (numerics A/B), everything else = full v3
extern "C" void ds_flash_fwd_dbg(const void* q, const void* k, const void* vt,
                                 void* o, int B, int S, int H, int Hkv,
                                 float scale, int variant, void* stream) {
  dim3 grid((S + QTILE - 1) / QTILE, H, B);
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (variant == 0)
    hipLaunchKernelGGL((flash_fwd_kernel<true, false>), grid, dim3(TPB), 0,
                       st, (const short*)q, (const short*)k, (const short*)vt,
                       (short*)o, nullptr, B, S, H, Hkv, scale);
  else
    hipLaunchKernelGGL((flash_fwd_kernel<true, true>), grid, dim3(TPB), 0, st,
                       (const short*)q, (const short*)k, (const short*)vt,
                       (short*)o, nullptr, B, S, H, Hkv, scale);
}
