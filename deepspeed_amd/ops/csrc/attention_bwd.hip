// Flash-attention backward for MI355X (gfx950 / CDNA4) — hand-written MFMA.
//
// v2: 8-wave workgroups with LDS-shared operand tiles. v1 ran one wave per
// 32-row tile and re-read Q/dO (or K/V) from global for every tile pair —
// ~30 flops/byte, HBM-bound at ~110 TF. v2 blocks 256 rows per workgroup
// (8 waves x 32) and stages the per-iteration operand tiles in
// double-buffered LDS shared by all 8 waves: ~260 flops/byte, compute-
// bound. Same math as v1 (tile-for-tile the CPU-validated blueprint
// ops/flash_bwd_ref.py; GPU-numerics-validated vs torch autograd).
//
// Layouts (ALL [B, H(kv), S, D] = BHSD, contiguous):
//   q, o, do: [B, H, S, D]      k, v: [B, Hkv, S, D]
//   qt, kt, dot: transposed copies [B, H(kv), D, S] (wrapper-made) so the
//   dK/dV/dQ MFMAs' B-operands (k-dim = q or kv) stage contiguously.
//   lse, delta: [B, H, S] fp32.  dq/dk/dv bf16 outputs.
//
// MFMA fragment maps (probe-verified, scripts/mfma_probe.hip):
//   A: row=lane&31, k=reg+8*(lane>>5) · B: col=lane&31, same k
//   C/D: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
//
// dkdv kernel: grid (S/256, Hkv, B). Each wave owns 32 kv rows (K/V
// A-fragments preloaded in registers; dk/dv accumulate in registers across
// the whole q loop — the GQA head group G=H/Hkv is an inner loop, no
// atomics). Per (g, q-tile) iteration the block stages Q[32][128],
// dO[32][128], qt[128][32], dot[128][32] (32 KB) double-buffered with the
// async issue-early/write-late split and ONE barrier per iteration.
//
// dq kernel: grid (S/256, H, B). Each wave owns 32 q rows (Q/dO
// B-fragments + dq accumulator in registers); stages K[32][128],
// V[32][128], kt[128][32] per kv tile.
//
// The C/D -> A-operand change of axis for P^T / dS^T goes through a
// per-wave 32x32 bf16 LDS scratch (XOR-swizzled 16B chunks), no barrier
// needed (same-wave LDS RAW is ordered by lgkmcnt).

#include "ds_kernels.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8s;
typedef __attribute__((ext_vector_type(8))) short lds_chunk;  // 16 B
typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int T = 32;     // tile rows
constexpr int D = 128;    // head dim
constexpr int NW = 8;     // waves per workgroup
constexpr int MT = T * NW;  // macro tile (256 rows)
constexpr int TPB = NW * 64;

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

// [32][128] tiles: 16 chunks/row, XOR over row&7 (fwd-verified pattern)
__device__ __forceinline__ int sw16(int row, int chunk) {
  return row * 16 + (chunk ^ (row & 7));
}
// [128][32] tiles and 32x32 scratch: 4 chunks/row, XOR over row&3
__device__ __forceinline__ int sw4(int row, int chunk) {
  return row * 4 + (chunk ^ (row & 3));
}
// [128][64] tiles: 8 chunks/row, XOR over row&7
__device__ __forceinline__ int sw8(int row, int chunk) {
  return row * 8 + (chunk ^ (row & 7));
}

// scalar bf16 store into a sw4-swizzled [32][32] scratch
__device__ __forceinline__ void scr_store(short* scr, int row, int col,
                                          short v) {
  const int chunk = (col >> 3) ^ (row & 3);
  scr[row * 32 + chunk * 8 + (col & 7)] = v;
}
// 8-element read at (row, k0=kk*16+8*half) from the same scratch
__device__ __forceinline__ bf16x8s scr_read(const short* scr, int row,
                                            int kk, int half) {
  const int chunk = (2 * kk + half) ^ (row & 3);
  return *(const bf16x8s*)(scr + row * 32 + chunk * 8);
}

// ---------------------------------------------------------------- dk/dv

// Staging via global_load_lds: the LDS destination is linear
// (wave-uniform base + lane*16, the hardware contract), and the SWIZZLE is
// folded into the per-lane GLOBAL source address (inverse permutation) —
// zero staging registers, fully async (vmcnt-counted, drained by the
// barrier). Each thread issues one 16 B chunk per tile kind.
__device__ __forceinline__ void dkdv_stage(
    const short* __restrict__ q, const short* __restrict__ dout,
    const short* __restrict__ qt, const short* __restrict__ dot,
    lds_chunk* qb, lds_chunk* dob, lds_chunk* qtb, lds_chunk* dotb,
    long long qbase, long long tbase, int qs, int S, int tid) {
#pragma unroll
  for (int t = 0; t < 2; ++t) {
    const int j = tid + t * TPB;    // linear LDS chunk slot 0..1023
    // [64][128] tiles (16 chunks/row): slot j = (row=j>>4, c=(j&15)^(row&7))
    const int row = j >> 4, c = (j & 15) ^ (row & 7);
    const long long qoff =
        qbase + (long long)min(qs + row, S - 1) * D + c * 8;
    __builtin_amdgcn_global_load_lds((const unsigned int*)(q + qoff),
                                     (unsigned int*)(qb + j), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((const unsigned int*)(dout + qoff),
                                     (unsigned int*)(dob + j), 16, 0, 0);
    // [128][64] tiles (8 chunks/row): slot j = (vr=j>>3, vc=(j&7)^(vr&7))
    const int vr = j >> 3, vc = (j & 7) ^ (vr & 7);
    const long long toff = tbase + (long long)vr * S +
                           min(qs + vc * 8, S - 8);
    __builtin_amdgcn_global_load_lds((const unsigned int*)(qt + toff),
                                     (unsigned int*)(qtb + j), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((const unsigned int*)(dot + toff),
                                     (unsigned int*)(dotb + j), 16, 0, 0);
  }
}

// dkdv v3 geometry: KV macro-tile = 128 rows = 4 wave-PAIRS x 32 rows.
// Both waves of a pair cover the same 32 kv rows (K/V staged once per
// pair in LDS, read as A-fragments like the dq kernel) and split the d
// dimension: wave pm=0 accumulates dblk 0-1, pm=1 dblk 2-3 — so the
// dk/dv accumulators are 64 VGPRs per wave instead of 128 (the
// register-resident variants spilled 100-268 B/lane and the per-
// iteration global A-fragment loads serialized against the MFMAs,
// r2_call6 ablation).
constexpr int MTK = 128;  // kv rows per workgroup

template <bool CAUSAL, int VAR = 0>
__global__ __launch_bounds__(TPB) void flash_bwd_dkdv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const short* __restrict__ qt, const short* __restrict__ dot,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv,
    const int B, const int S, const int H, const int Hkv,
    const float scale) {
  __shared__ lds_chunk kscr[4][T * 16];   // K [32][128] per pair (32 KB)
  __shared__ lds_chunk vscr[4][T * 16];   // V (32 KB)
  __shared__ lds_chunk qbuf[2 * T * 16];  // single-buffered 64-row q tiles
  __shared__ lds_chunk dobuf[2 * T * 16];
  __shared__ lds_chunk qtbuf[D * 8];      // [128][64]
  __shared__ lds_chunk dotbuf[D * 8];     // (64 KB of tiles total)
  __shared__ short pscr[NW][T * T];       // per-wave scratch, REUSED for
                                          // pt then dst per subtile (16 KB)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int pair = wid >> 1;
  const int pm = wid & 1;                // my d half: dblk = pm*2 + {0,1}
  const int col = lane & 31;
  const int half = lane >> 5;
  const int kv0b = blockIdx.x * MTK;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int G = H / Hkv;
  const int my_kv0 = kv0b + pair * T;
  const int kvrow = my_kv0 + col;

  // ---- stage K/V for all 4 pairs (linear dst, pre-swizzled src).
  // k/v are BHSD here ([B,Hkv,S,D]) — kv-row stride is D, unlike the
  // forward kernel's [B,S,Hkv,D] layout.
  {
    const long long kb0 = (((long long)b * Hkv + hkv) * S) * D;
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const int j = tid + t * TPB;       // 0..2047
      const int p = j >> 9, jj = j & 511;
      const int row = jj >> 4, c = (jj & 15) ^ (row & 7);
      const long long off =
          kb0 + (long long)min(kv0b + p * T + row, S - 1) * D + c * 8;
      __builtin_amdgcn_global_load_lds((const unsigned int*)(k + off),
                                       (unsigned int*)(&kscr[0][0] + j),
                                       16, 0, 0);
      __builtin_amdgcn_global_load_lds((const unsigned int*)(v + off),
                                       (unsigned int*)(&vscr[0][0] + j),
                                       16, 0, 0);
    }
  }

  f32x16 dvacc[2] = {};  // my two 32-wide d blocks
  f32x16 dkacc[2] = {};

  const int QT = 2 * T;                    // 64 q rows per iteration
  const int q_start = CAUSAL ? kv0b : 0;   // 64-aligned (kv0b % 128 == 0)
  const int nq = (S - q_start + QT - 1) / QT;
  const int total = G * nq;

  auto qb_of = [&](int it, int& qs, long long& qbase, long long& tbase,
                   long long& sbase) {
    const int g = it / nq;
    qs = q_start + (it % nq) * (2 * T);
    const int h = hkv * G + g;
    qbase = (((long long)b * H + h) * S) * D;
    tbase = (((long long)b * H + h) * D) * S;
    sbase = ((long long)b * H + h) * S;
  };

  {
    int qs;
    long long qbase, tbase, sbase;
    qb_of(0, qs, qbase, tbase, sbase);
    dkdv_stage(q, dout, qt, dot, qbuf, dobuf, qtbuf, dotbuf,
               qbase, tbase, qs, S, tid);
  }
  __syncthreads();

  short* pw = pscr[wid];
  const lds_chunk* kp = kscr[pair];
  const lds_chunk* vp = vscr[pair];

  for (int it = 0; it < total; ++it) {
    int qs;
    long long qbase, tbase, sbase;
    qb_of(it, qs, qbase, tbase, sbase);

    if (!(VAR & 1)) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {       // two 32-row q subtiles
        const int qs2 = qs + sub * T;
        if (CAUSAL && qs2 + T - 1 < my_kv0) continue;
        // ---- S^T = K Q^T ; dP^T = V dO^T (A from pair LDS, B from tiles)
        f32x16 st = {}, dpt = {};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          const bf16x8s kA = kp[sw16(col, kk * 2 + half)];
          const bf16x8s qB = qbuf[sw16(sub * T + col, kk * 2 + half)];
          st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kA, qB, st, 0, 0, 0);
        }
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          const bf16x8s vA = vp[sw16(col, kk * 2 + half)];
          const bf16x8s doB = dobuf[sw16(sub * T + col, kk * 2 + half)];
          dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vA, doB, dpt,
                                                        0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);

        const int qrow = qs2 + col;
        const float l = lse[sbase + min(qrow, S - 1)];
        const float dl = delta[sbase + min(qrow, S - 1)];
        if (VAR & 2) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {  // keep st/dpt live, skip the rest
            dvacc[0][r] += st[r] * l;
            dkacc[0][r] += dpt[r] * dl;
          }
          continue;
        }
        // pt pass: write P^T into the (single, reused) scratch
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kvl = cd_row(r, half);  // kv row within my pair tile
          const bool dead = (CAUSAL && (my_kv0 + kvl > qrow)) ||
                            (my_kv0 + kvl >= S) || (qrow >= S);
          const float pt = dead ? 0.f : __expf(st[r] * scale - l);
          st[r] = pt;                       // keep pt for the dst pass
          scr_store(pw, kvl, col, f2bf(pt));
        }
        // ---- dV[kv][d] += P^T(k=q) dO-asB
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int db = 0; db < 2; ++db) {
          const int dblk = pm * 2 + db;
#pragma unroll
          for (int kk = 0; kk < 2; ++kk) {
            const bf16x8s pA = scr_read(pw, col, kk, half);
            const bf16x8s doB =
                dotbuf[sw8(dblk * 32 + col, sub * 4 + kk * 2 + half)];
            dvacc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                pA, doB, dvacc[db], 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);
        // dst pass: overwrite the scratch (same-wave LDS RAW is ordered)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kvl = cd_row(r, half);
          const float dst = st[r] * (dpt[r] - dl) * scale;
          scr_store(pw, kvl, col, f2bf(dst));
        }
        // ---- dK[kv][d] += dS^T(k=q) Q-asB
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int db = 0; db < 2; ++db) {
          const int dblk = pm * 2 + db;
#pragma unroll
          for (int kk = 0; kk < 2; ++kk) {
            const bf16x8s dA = scr_read(pw, col, kk, half);
            const bf16x8s qB2 =
                qtbuf[sw8(dblk * 32 + col, sub * 4 + kk * 2 + half)];
            dkacc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                dA, qB2, dkacc[db], 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);
      }
    }

    __syncthreads();  // everyone done reading the tile buffers
    const bool have_next = (it + 1 < total) && !(VAR & 4);
    if (have_next) {
      int qs2;
      long long qb2, tb2, sb2;
      qb_of(it + 1, qs2, qb2, tb2, sb2);
      dkdv_stage(q, dout, qt, dot, qbuf, dobuf, qtbuf, dotbuf,
                 qb2, tb2, qs2, S, tid);
    }
    __syncthreads();  // loads drained (barrier waits vmcnt)
  }

  // ---- write dk/dv for my 2 d blocks (col = d_local, row = kv via map)
  const long long obase = (((long long)b * Hkv + hkv) * S) * D;
#pragma unroll
  for (int db = 0; db < 2; ++db) {
    const int dblk = pm * 2 + db;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kvl = cd_row(r, half);
      if (my_kv0 + kvl >= S) continue;
      const long long off = obase + (long long)(my_kv0 + kvl) * D +
                            dblk * 32 + col;
      dv[off] = f2bf(dvacc[db][r]);
      dk[off] = f2bf(dkacc[db][r]);
    }
  }
}


// ------------------------------------------------------------------- dq

struct DqStage {
  bf16x8s kv_, vv, ktv;
};

__device__ __forceinline__ DqStage dq_load(const short* __restrict__ k,
                                           const short* __restrict__ v,
                                           const short* __restrict__ kt,
                                           long long kbase, long long ktbase,
                                           int kv0, int S, int tid) {
  DqStage r;
  const int row = tid >> 4, c = tid & 15;
  const long long koff = kbase + (long long)min(kv0 + row, S - 1) * D + c * 8;
  r.kv_ = *(const bf16x8s*)(k + koff);
  r.vv = *(const bf16x8s*)(v + koff);
  const int vr = tid >> 2, vc = tid & 3;
  const long long toff = ktbase + (long long)vr * S +
                         min(kv0 + vc * 8, S - 8);
  r.ktv = *(const bf16x8s*)(kt + toff);
  return r;
}

__device__ __forceinline__ void dq_write(const DqStage& r, lds_chunk* kb,
                                         lds_chunk* vb, lds_chunk* ktb,
                                         int tid) {
  const int row = tid >> 4, c = tid & 15;
  kb[sw16(row, c)] = r.kv_;
  vb[sw16(row, c)] = r.vv;
  const int vr = tid >> 2, vc = tid & 3;
  ktb[sw4(vr, vc)] = r.ktv;
}

template <bool CAUSAL>
__global__ __launch_bounds__(TPB) void flash_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const short* __restrict__ kt,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq,
    const int B, const int S, const int H, const int Hkv,
    const float scale) {
  __shared__ lds_chunk kbuf[2][T * 16];
  __shared__ lds_chunk vbuf[2][T * 16];
  __shared__ lds_chunk ktbuf[2][D * 4];
  __shared__ short dscr2[NW][T * T];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;
  const int half = lane >> 5;
  const int q0b = blockIdx.x * MT;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (H / Hkv);
  const int q0w = q0b + wid * T;
  const int qrow = q0w + col;
  const int qload = min(qrow, S - 1);

  const long long qbase = (((long long)b * H + h) * S) * D;
  const long long kbase = (((long long)b * Hkv + hkv) * S) * D;
  const long long ktbase = (((long long)b * Hkv + hkv) * D) * S;
  const long long sbase = ((long long)b * H + h) * S;

  // ---- B-frags of Q and dO for my q rows
  bf16x8s qf[8], dof[8];
#pragma unroll
  for (int kk = 0; kk < 8; ++kk) {
    qf[kk] = *(const bf16x8s*)(q + qbase + (long long)qload * D + kk * 16 +
                               8 * half);
    dof[kk] = *(const bf16x8s*)(dout + qbase + (long long)qload * D +
                                kk * 16 + 8 * half);
  }
  const float l = lse[sbase + qload];
  const float dl = delta[sbase + qload];

  f32x16 dqacc[4] = {};  // rows=q, cols=d (col=lane&31=d_local)

  const int kv_end_blk = CAUSAL ? min(S, q0b + MT) : S;
  const int kv_end_wave = CAUSAL ? min(S, q0w + T) : S;
  const int nt = (kv_end_blk + T - 1) / T;

  {
    DqStage sr = dq_load(k, v, kt, kbase, ktbase, 0, S, tid);
    dq_write(sr, kbuf[0], vbuf[0], ktbuf[0], tid);
  }
  __syncthreads();

  short* dw = dscr2[wid];

  for (int it = 0; it < nt; ++it) {
    const int kv0 = it * T;
    const int cur = it & 1;
    DqStage sr;
    const bool have_next = (it + 1 < nt);
    if (have_next)
      sr = dq_load(k, v, kt, kbase, ktbase, kv0 + T, S, tid);

    if (kv0 < kv_end_wave) {
      const lds_chunk* kb = kbuf[cur];
      const lds_chunk* vb = vbuf[cur];
      const lds_chunk* ktb = ktbuf[cur];

      f32x16 st = {}, dpt = {};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        const bf16x8s kA = kb[sw16(col, kk * 2 + half)];
        const bf16x8s vA = vb[sw16(col, kk * 2 + half)];
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kA, qf[kk], st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vA, dof[kk], dpt,
                                                      0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvl = cd_row(r, half);
        const bool dead = (CAUSAL && (kv0 + kvl > qrow)) ||
                          (kv0 + kvl >= S);
        const float pt = dead ? 0.f : __expf(st[r] * scale - l);
        const float dst = pt * (dpt[r] - dl) * scale;
        // store TRANSPOSED: [q][kv] so dS reads as A-frags (rows q)
        scr_store(dw, col, kvl, f2bf(dst));
      }

      // ---- dQ[q][d] += dS(row=q, k=kv) · K^T-asB(k=kv, col=d)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dblk = 0; dblk < 4; ++dblk) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const bf16x8s dA = scr_read(dw, col, kk, half);
          const bf16x8s kB = ktb[sw4(dblk * 32 + col, kk * 2 + half)];
          dqacc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              dA, kB, dqacc[dblk], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }

    if (have_next)
      dq_write(sr, kbuf[cur ^ 1], vbuf[cur ^ 1], ktbuf[cur ^ 1], tid);
    __syncthreads();
  }

  // ---- write dq (rows=q via reg map, col=d_local)
  if (q0w >= S) return;
#pragma unroll
  for (int dblk = 0; dblk < 4; ++dblk) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int ql = cd_row(r, half);
      if (q0w + ql >= S) continue;
      dq[qbase + (long long)(q0w + ql) * D + dblk * 32 + col] =
          f2bf(dqacc[dblk][r]);
    }
  }
}


}  // namespace

// probe-only ablation entry (scripts/bwd_ablate.hip)
extern "C" void ds_flash_bwd_dkdv_dbg(const void* q, const void* k,
                                      const void* v, const void* dout,
                                      const void* qt, const void* dot,
                                      const float* lse, const float* delta,
                                      void* dk, void* dv, int B, int S,
                                      int H, int Hkv, float scale,
                                      int variant, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  dim3 g1((S + MT - 1) / MT, Hkv, B);
#define L(V) hipLaunchKernelGGL((flash_bwd_dkdv_kernel<true, V>), g1, \
      dim3(TPB), 0, st, (const short*)q, (const short*)k, (const short*)v, \
      (const short*)dout, (const short*)qt, (const short*)dot, lse, delta, \
      (short*)dk, (short*)dv, B, S, H, Hkv, scale)
  switch (variant) {
    case 1: L(1); break;
    case 2: L(2); break;
    case 4: L(4); break;
    case 5: L(5); break;
    default: L(0);
  }
#undef L
}

extern "C" void ds_flash_bwd(const void* q, const void* k, const void* v,
                             const void* dout, const void* qt,
                             const void* kt, const void* dot,
                             const float* lse, const float* delta, void* dq,
                             void* dk, void* dv, int B, int S, int H,
                             int Hkv, float scale, int causal, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  dim3 g1((S + MTK - 1) / MTK, Hkv, B);
  dim3 g2((S + MT - 1) / MT, H, B);
  if (causal) {
    hipLaunchKernelGGL((flash_bwd_dkdv_kernel<true>), g1, dim3(TPB), 0, st,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const short*)qt,
                       (const short*)dot, lse, delta, (short*)dk, (short*)dv,
                       B, S, H, Hkv, scale);
    hipLaunchKernelGGL((flash_bwd_dq_kernel<true>), g2, dim3(TPB), 0, st,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const short*)kt, lse, delta,
                       (short*)dq, B, S, H, Hkv, scale);
  } else {
    hipLaunchKernelGGL((flash_bwd_dkdv_kernel<false>), g1, dim3(TPB), 0, st,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const short*)qt,
                       (const short*)dot, lse, delta, (short*)dk, (short*)dv,
                       B, S, H, Hkv, scale);
    hipLaunchKernelGGL((flash_bwd_dq_kernel<false>), g2, dim3(TPB), 0, st,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const short*)kt, lse, delta,
                       (short*)dq, B, S, H, Hkv, scale);
  }
}
