// Paged flash-decode attention for MI355X (gfx950).
//
// Reference analogue: inference/v2/kernels/ragged_ops/blocked_flash
// (flash_attn_by_atoms over paged KV). Single-token decode is HBM-bound
// KV streaming (guide §B "Attention decode"), so this is a split-S
// flash-decode over the block table, no MFMA:
//
//   kernel 1 (partials): grid (n_seqs, Hkv, S_SPLITS). One workgroup =
//     4 waves; each wave owns a slice of the split's positions. A wave
//     computes ALL G = H/Hkv query heads of its kv head together (K/V
//     read ONCE for the whole GQA group — the whole point of GQA).
//     Per position: cooperative dot (lane owns 2 of D=128 elems),
//     wave-reduce per head, online-softmax accumulate into 2 f32/lane
//     per head. Wave partials combine in LDS, then one (m, l, acc[D])
//     partial per (seq, kv head, split, q head) goes to global.
//   kernel 2 (combine): one wave per (seq, head): logsumexp-merge the
//     split partials into o [n, H, D].
//
// Layouts: q [n, H, D] bf16; pool k/v [nblocks, Hkv, BS, D] bf16;
// block_table [n, max_blocks] int32; lens [n] int32. D = 128, BS = any
// multiple of 1. Output o [n, H, D] bf16.

#include "ds_kernels.h"

namespace {

constexpr int D = 128;
constexpr int NW = 4;  // waves per workgroup (kernel 1)
constexpr int MAXG = 8;

typedef __attribute__((ext_vector_type(2))) float f32x2;

template <int G>
__global__ __launch_bounds__(NW * 64) void paged_decode_partial_kernel(
    const short* __restrict__ q,        // [n, H, D]
    const short* __restrict__ kpool,    // [nb, Hkv, BS, D]
    const short* __restrict__ vpool,
    const int* __restrict__ block_table,  // [n, max_blocks]
    const int* __restrict__ lens,         // [n]
    float* __restrict__ part,   // [n, Hkv, SPLITS, G, D] accumulators
    float* __restrict__ part_ml,  // [n, Hkv, SPLITS, G, 2] (m, l)
    const int H, const int Hkv, const int BS, const int max_blocks,
    const int splits, const float scale) {
  const int seq = blockIdx.x;
  const int hkv = blockIdx.y;
  const int split = blockIdx.z;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;

  const int len = lens[seq];
  const int per_split = (len + splits - 1) / splits;
  const int p0 = split * per_split;
  const int p1 = min(len, p0 + per_split);

  // q fragments for my kv head's G query heads: lane owns d = 2*lane, +1
  float qf[G][2];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const int h = hkv * G + g;
    const short* qp = q + ((long long)seq * H + h) * D + 2 * lane;
    qf[g][0] = ds::to_f32(*(const ds::bf16*)(qp));
    qf[g][1] = ds::to_f32(*(const ds::bf16*)(qp + 1));
  }

  float m[G], l[G], acc[G][2];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = -1e30f;
    l[g] = 0.f;
    acc[g][0] = acc[g][1] = 0.f;
  }

  const int* bt = block_table + (long long)seq * max_blocks;
  // wave w handles positions p0 + w, stride NW
  for (int p = p0 + wid; p < p1; p += NW) {
    const int blk = bt[p / BS];
    const long long base =
        (((long long)blk * Hkv + hkv) * BS + (p % BS)) * D + 2 * lane;
    const float k0 = ds::to_f32(*(const ds::bf16*)(kpool + base));
    const float k1 = ds::to_f32(*(const ds::bf16*)(kpool + base + 1));
    const float v0 = ds::to_f32(*(const ds::bf16*)(vpool + base));
    const float v1 = ds::to_f32(*(const ds::bf16*)(vpool + base + 1));
    float s[G];
#pragma unroll
    for (int g = 0; g < G; ++g) s[g] = qf[g][0] * k0 + qf[g][1] * k1;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
#pragma unroll
      for (int g = 0; g < G; ++g) s[g] += __shfl_xor(s[g], off, 64);
    }
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float sc = s[g] * scale;
      const float mn = fmaxf(m[g], sc);
      const float alpha = __expf(m[g] - mn);
      const float p_ = __expf(sc - mn);
      l[g] = l[g] * alpha + p_;
      acc[g][0] = acc[g][0] * alpha + p_ * v0;
      acc[g][1] = acc[g][1] * alpha + p_ * v1;
      m[g] = mn;
    }
  }

  // ---- combine the NW waves' partials through LDS (logsumexp merge)
  __shared__ float lds_acc[NW][MAXG][D];
  __shared__ float lds_ml[NW][MAXG][2];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    lds_acc[wid][g][2 * lane] = acc[g][0];
    lds_acc[wid][g][2 * lane + 1] = acc[g][1];
    if (lane == 0) {
      lds_ml[wid][g][0] = m[g];
      lds_ml[wid][g][1] = l[g];
    }
  }
  __syncthreads();
  if (wid == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float mg = -1e30f;
      for (int w = 0; w < NW; ++w) mg = fmaxf(mg, lds_ml[w][g][0]);
      float lg = 0.f, a0 = 0.f, a1 = 0.f;
      for (int w = 0; w < NW; ++w) {
        const float al = __expf(lds_ml[w][g][0] - mg);
        lg += lds_ml[w][g][1] * al;
        a0 += lds_acc[w][g][2 * lane] * al;
        a1 += lds_acc[w][g][2 * lane + 1] * al;
      }
      const long long ob =
          ((((long long)seq * Hkv + hkv) * gridDim.z + split) * G + g);
      part[ob * D + 2 * lane] = a0;
      part[ob * D + 2 * lane + 1] = a1;
      if (lane == 0) {
        part_ml[ob * 2] = mg;
        part_ml[ob * 2 + 1] = lg;
      }
    }
  }
}

__global__ __launch_bounds__(64) void paged_decode_combine_kernel(
    const float* __restrict__ part,      // [n, Hkv, SPLITS, G, D]
    const float* __restrict__ part_ml,   // [n, Hkv, SPLITS, G, 2]
    short* __restrict__ o,               // [n, H, D]
    const int H, const int Hkv, const int splits) {
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int lane = threadIdx.x;
  const int G = H / Hkv;
  const int hkv = h / G, g = h % G;
  float mg = -1e30f;
  for (int s = 0; s < splits; ++s) {
    const long long ob = ((((long long)seq * Hkv + hkv) * splits + s) * G + g);
    mg = fmaxf(mg, part_ml[ob * 2]);
  }
  float lg = 0.f, a0 = 0.f, a1 = 0.f;
  for (int s = 0; s < splits; ++s) {
    const long long ob = ((((long long)seq * Hkv + hkv) * splits + s) * G + g);
    const float al = __expf(part_ml[ob * 2] - mg);
    lg += part_ml[ob * 2 + 1] * al;
    a0 += part[ob * D + 2 * lane] * al;
    a1 += part[ob * D + 2 * lane + 1] * al;
  }
  const float inv = lg > 0.f ? 1.f / lg : 0.f;
  ds::bf16* op = (ds::bf16*)(o + ((long long)seq * H + h) * D);
  op[2 * lane] = ds::from_f32<ds::bf16>(a0 * inv);
  op[2 * lane + 1] = ds::from_f32<ds::bf16>(a1 * inv);
}

}  // namespace

extern "C" void ds_paged_decode(const void* q, const void* kpool,
                                const void* vpool, const int* block_table,
                                const int* lens, float* part, float* part_ml,
                                void* o, int n, int H, int Hkv, int BS,
                                int max_blocks, int splits, float scale,
                                void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const int G = H / Hkv;
  dim3 g1(n, Hkv, splits);
#define L(G_)                                                                 \
  hipLaunchKernelGGL((paged_decode_partial_kernel<G_>), g1, dim3(NW * 64), 0, \
                     st, (const short*)q, (const short*)kpool,                \
                     (const short*)vpool, block_table, lens, part, part_ml,   \
                     H, Hkv, BS, max_blocks, splits, scale)
  switch (G) {
    case 1: L(1); break;
    case 2: L(2); break;
    case 4: L(4); break;
    case 8: L(8); break;
    default: return;  // wrapper guards G in {1,2,4,8}
  }
#undef L
  dim3 g2(n, H);
  hipLaunchKernelGGL(paged_decode_combine_kernel, g2, dim3(64), 0, st, part,
                     part_ml, (short*)o, H, Hkv, splits);
}
