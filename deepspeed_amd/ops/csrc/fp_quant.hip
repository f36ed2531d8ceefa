// Group-wise low-bit FLOAT quantization for MI355X (gfx950).
//
// Reference analogue: csrc/fp_quantizer/fp_quantize.cu (FP4/FP6/FP8/FP12
// quantize/dequantize with per-group scales). Re-designed rather than
// ported: one wave per group, wave-reduce absmax, RNE bit-manipulation
// conversion, TIGHT packing (FP6: 4 values -> 3 bytes; FP12: 2 -> 3;
// FP4: 2 -> 1; FP8: 1 -> 1), scales stored separately as fp32.
//
// Formats (sign + exponent + mantissa):
//   FP4  = e2m1,  FP6 = e3m2,  FP8 = e4m3,  FP12 = e7m4
// (matching the reference's q_bits -> mantissa mapping, quantize.py:71-77).
// The per-group scale maps the group's absmax onto the format's max
// normal value, so the exponent range is centered on the data.

#include "ds_kernels.h"

namespace {

template <int E, int M>
__device__ __forceinline__ unsigned f32_to_fp(float x) {
  // RNE conversion of x into sign|E|M bits (saturating, subnormal-aware)
  union {
    float f;
    unsigned u;
  } c;
  c.f = x;
  const unsigned sign = c.u >> 31;
  int exp = (int)((c.u >> 23) & 255) - 127;       // unbiased
  unsigned man = c.u & 0x7fffff;
  const int bias = (1 << (E - 1)) - 1;
  const int emax = (1 << E) - 2 - bias;           // max normal exponent
  const int emin = 1 - bias;                      // min normal exponent
  if ((c.u & 0x7fffffff) == 0) return sign << (E + M);
  // round mantissa to M bits (RNE)
  const int shift = 23 - M;
  unsigned keep = man >> shift;
  const unsigned rem = man & ((1u << shift) - 1);
  const unsigned halfway = 1u << (shift - 1);
  if (rem > halfway || (rem == halfway && (keep & 1))) {
    keep += 1;
    if (keep == (1u << M)) {  // mantissa overflow -> bump exponent
      keep = 0;
      exp += 1;
    }
  }
  if (exp > emax) {  // saturate to max finite
    return (sign << (E + M)) | (((1u << E) - 2) << M) | ((1u << M) - 1);
  }
  if (exp < emin) {  // subnormal: value = man_total * 2^(emin - M)
    const float scale = __builtin_exp2f((float)(M - emin));
    float mag = fabsf(x) * scale;                 // in units of 2^(emin-M)
    unsigned q = (unsigned)rintf(mag);
    if (q > ((1u << M) - 1)) {                    // rounded up to normal min
      return (sign << (E + M)) | (1u << M);
    }
    return (sign << (E + M)) | q;
  }
  return (sign << (E + M)) | ((unsigned)(exp + bias) << M) | keep;
}

template <int E, int M>
__device__ __forceinline__ float fp_to_f32(unsigned q) {
  const unsigned sign = (q >> (E + M)) & 1;
  const unsigned eb = (q >> M) & ((1u << E) - 1);
  const unsigned mb = q & ((1u << M) - 1);
  const int bias = (1 << (E - 1)) - 1;
  float v;
  if (eb == 0) {
    v = (float)mb * __builtin_exp2f((float)(1 - bias - M));
  } else {
    v = (1.f + (float)mb / (float)(1 << M)) *
        __builtin_exp2f((float)((int)eb - bias));
  }
  return sign ? -v : v;
}

template <int E, int M>
__device__ __forceinline__ float fp_max() {
  const int bias = (1 << (E - 1)) - 1;
  return (2.f - 1.f / (float)(1 << M)) *
         __builtin_exp2f((float)(((1 << E) - 2) - bias));
}

// one wave per group; BITS in {4, 6, 8, 12}; group_size % 8 == 0
template <typename T, int E, int M, int BITS>
__global__ void fp_quant_kernel(const T* __restrict__ x,
                                unsigned char* __restrict__ out,
                                float* __restrict__ scales,
                                const long long n, const int group_size) {
  const long long g = (long long)blockIdx.x * (blockDim.x >> 6) +
                      (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const long long g0 = g * group_size;
  if (g0 >= n) return;
  const int len = (int)min((long long)group_size, n - g0);

  float amax = 0.f;
  for (int i = lane; i < len; i += 64)
    amax = fmaxf(amax, fabsf(ds::to_f32(x[g0 + i])));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, 64));
  const float qmax = fp_max<E, M>();
  const float scale = amax > 0.f ? amax / qmax : 1.f;
  const float inv = 1.f / scale;
  if (lane == 0) scales[g] = scale;

  // pack: each lane handles 24 bits = LCM-friendly unit (values_per_3B:
  // BITS=4 -> 6, 6 -> 4, 8 -> 3, 12 -> 2). Groups are byte-padded to a
  // whole number of 3-byte units so group boundaries never split a unit
  // (group_size need not divide vper3).
  const int vper3 = 24 / BITS;
  const int gu = (group_size + vper3 - 1) / vper3;  // units per group
  const int n3 = (len + vper3 - 1) / vper3;         // units in THIS group
  unsigned char* gout = out + (long long)g * gu * 3;
  for (int u = lane; u < n3; u += 64) {
    unsigned word = 0;
    for (int t = 0; t < vper3; ++t) {
      const int idx = u * vper3 + t;
      const float v = idx < len ? ds::to_f32(x[g0 + idx]) * inv : 0.f;
      word |= f32_to_fp<E, M>(v) << (t * BITS);
    }
    gout[u * 3 + 0] = word & 255;
    gout[u * 3 + 1] = (word >> 8) & 255;
    gout[u * 3 + 2] = (word >> 16) & 255;
  }
}

template <typename T, int E, int M, int BITS>
__global__ void fp_dequant_kernel(const unsigned char* __restrict__ qd,
                                  const float* __restrict__ scales,
                                  T* __restrict__ out, const long long n,
                                  const int group_size) {
  const long long g = (long long)blockIdx.x * (blockDim.x >> 6) +
                      (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const long long g0 = g * group_size;
  if (g0 >= n) return;
  const int len = (int)min((long long)group_size, n - g0);
  const float scale = scales[g];
  const int vper3 = 24 / BITS;
  const int gu = (group_size + vper3 - 1) / vper3;
  const int n3 = (len + vper3 - 1) / vper3;
  const unsigned char* gin = qd + (long long)g * gu * 3;
  for (int u = lane; u < n3; u += 64) {
    unsigned word = (unsigned)gin[u * 3] | ((unsigned)gin[u * 3 + 1] << 8) |
                    ((unsigned)gin[u * 3 + 2] << 16);
    for (int t = 0; t < vper3; ++t) {
      const int idx = u * vper3 + t;
      if (idx >= len) break;
      const unsigned q = (word >> (t * BITS)) & ((1u << BITS) - 1);
      out[g0 + idx] = ds::from_f32<T>(fp_to_f32<E, M>(q) * scale);
    }
  }
}

template <typename T>
void launch_all(const void* x, void* out, float* scales, long long n,
                int group_size, int bits, bool dequant, hipStream_t st) {
  const long long groups = (n + group_size - 1) / group_size;
  const int waves_per_block = 4;
  const int block = waves_per_block * 64;
  const int grid = (int)((groups + waves_per_block - 1) / waves_per_block);
#define DISPATCH(E_, M_, B_)                                                  \
  if (dequant)                                                                \
    hipLaunchKernelGGL((fp_dequant_kernel<T, E_, M_, B_>), dim3(grid),        \
                       dim3(block), 0, st, (const unsigned char*)x, scales,   \
                       (T*)out, n, group_size);                               \
  else                                                                        \
    hipLaunchKernelGGL((fp_quant_kernel<T, E_, M_, B_>), dim3(grid),          \
                       dim3(block), 0, st, (const T*)x,                       \
                       (unsigned char*)out, scales, n, group_size)
  switch (bits) {
    case 4: DISPATCH(2, 1, 4); break;
    case 6: DISPATCH(3, 2, 6); break;
    case 8: DISPATCH(4, 3, 8); break;
    default: DISPATCH(7, 4, 12); break;
  }
#undef DISPATCH
}

}  // namespace

extern "C" void ds_fp_quantize(const void* x, int dtype, void* out,
                               float* scales, long long n, int group_size,
                               int bits, int dequant, void* stream) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (dtype == 0)
    launch_all<float>(x, out, scales, n, group_size, bits, dequant != 0, st);
  else if (dtype == 1)
    launch_all<ds::bf16>(x, out, scales, n, group_size, bits, dequant != 0,
                         st);
  else
    launch_all<ds::f16>(x, out, scales, n, group_size, bits, dequant != 0,
                        st);
}
