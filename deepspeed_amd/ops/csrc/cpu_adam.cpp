// CPU Adam/AdamW for ZeRO-Offload — host-side optimizer step over the
// pinned flat fp32 master shards.
//
// Capability parity with the reference's DeepSpeedCPUAdam
// (csrc/adam/cpu_adam_impl.cpp:22, csrc/includes/cpu_adam.h:110), different
// design: this framework's ZeRO optimizers keep ONE flat fp32 master buffer
// per param group, so the step is a single fused streaming pass — OpenMP
// across chunks, `omp simd` within (the compiler emits AVX2/AVX-512 FMA +
// vsqrtps; the loop is DRAM-bandwidth-bound at ~28 B/elem, so hand-written
// intrinsics buy nothing here). Optionally converts the updated params to
// bf16 in the same pass (the buffer the H2D copy ships back to HBM3E).

#include <atomic>
#include <cmath>
#include <cstdint>
#include <cstring>

#if defined(_OPENMP)
#include <omp.h>
#endif

namespace {

inline uint16_t f32_to_bf16_rne(float f) {
  uint32_t x;
  std::memcpy(&x, &f, 4);
  // round-to-nearest-even on the truncated 16 bits
  const uint32_t rounding = 0x7FFF + ((x >> 16) & 1);
  return (uint16_t)((x + rounding) >> 16);
}

inline float bf16_to_f32(uint16_t h) {
  uint32_t x = (uint32_t)h << 16;
  float f;
  std::memcpy(&f, &x, 4);
  return f;
}

inline float f16_to_f32(uint16_t h) {
  // scalar IEEE half -> float (grad arrays only; bandwidth-bound anyway)
  const uint32_t sign = (uint32_t)(h & 0x8000) << 16;
  uint32_t exp = (h >> 10) & 0x1F;
  uint32_t man = h & 0x3FF;
  uint32_t bits;
  if (exp == 0) {
    if (man == 0) {
      bits = sign;
    } else {  // subnormal
      exp = 127 - 15 + 1;
      while (!(man & 0x400)) {
        man <<= 1;
        --exp;
      }
      man &= 0x3FF;
      bits = sign | (exp << 23) | (man << 13);
    }
  } else if (exp == 0x1F) {
    bits = sign | 0x7F800000u | (man << 13);
  } else {
    bits = sign | ((exp + 127 - 15) << 23) | (man << 13);
  }
  float f;
  std::memcpy(&f, &bits, 4);
  return f;
}

template <typename GradLoad, bool ADAMW, bool WRITE_BF16>
void adam_loop(float* __restrict__ p, GradLoad gload, float* __restrict__ m,
               float* __restrict__ v, uint16_t* __restrict__ p16, int64_t n,
               float lr, float beta1, float beta2, float eps, float wd,
               float bc1, float bc2_sqrt, float inv_scale) {
  const float omb1 = 1.f - beta1;
  const float omb2 = 1.f - beta2;
#pragma omp parallel for schedule(static)
  for (int64_t c = 0; c < n; c += 4096) {
    const int64_t end = c + 4096 < n ? c + 4096 : n;
#pragma omp simd
    for (int64_t i = c; i < end; ++i) {
      float g = gload(i) * inv_scale;
      if (!ADAMW && wd != 0.f) g += wd * p[i];
      m[i] = beta1 * m[i] + omb1 * g;
      v[i] = beta2 * v[i] + omb2 * g * g;
      const float mhat = m[i] / bc1;
      const float denom = std::sqrt(v[i]) / bc2_sqrt + eps;
      float update = mhat / denom;
      if (ADAMW && wd != 0.f) update += wd * p[i];
      p[i] -= lr * update;
      if (WRITE_BF16) p16[i] = f32_to_bf16_rne(p[i]);
    }
  }
}

template <typename GradLoad>
void dispatch(float* p, GradLoad gload, float* m, float* v, uint16_t* p16,
              int64_t n, float lr, float b1, float b2, float eps, float wd,
              float bc1, float bc2s, float inv_scale, bool adamw) {
  if (adamw) {
    if (p16)
      adam_loop<GradLoad, true, true>(p, gload, m, v, p16, n, lr, b1, b2, eps,
                                      wd, bc1, bc2s, inv_scale);
    else
      adam_loop<GradLoad, true, false>(p, gload, m, v, p16, n, lr, b1, b2, eps,
                                       wd, bc1, bc2s, inv_scale);
  } else {
    if (p16)
      adam_loop<GradLoad, false, true>(p, gload, m, v, p16, n, lr, b1, b2, eps,
                                       wd, bc1, bc2s, inv_scale);
    else
      adam_loop<GradLoad, false, false>(p, gload, m, v, p16, n, lr, b1, b2,
                                        eps, wd, bc1, bc2s, inv_scale);
  }
}

}  // namespace

extern "C" void ds_cpu_adam_flat(float* p, const void* g, int grad_dtype,
                                 float* m, float* v, void* p16, long long n,
                                 float lr, float beta1, float beta2, float eps,
                                 float weight_decay, int step, float inv_scale,
                                 int adamw) {
  const float bc1 = 1.f - std::pow(beta1, (float)step);
  const float bc2s = std::sqrt(1.f - std::pow(beta2, (float)step));
  uint16_t* p16t = reinterpret_cast<uint16_t*>(p16);
  switch (grad_dtype) {
    case 0: {  // fp32
      const float* gf = reinterpret_cast<const float*>(g);
      dispatch(p, [gf](int64_t i) { return gf[i]; }, m, v, p16t, n, lr, beta1,
               beta2, eps, weight_decay, bc1, bc2s, inv_scale, adamw != 0);
      break;
    }
    case 1: {  // bf16
      const uint16_t* gb = reinterpret_cast<const uint16_t*>(g);
      dispatch(p, [gb](int64_t i) { return bf16_to_f32(gb[i]); }, m, v, p16t,
               n, lr, beta1, beta2, eps, weight_decay, bc1, bc2s, inv_scale,
               adamw != 0);
      break;
    }
    case 2: {  // fp16
      const uint16_t* gh = reinterpret_cast<const uint16_t*>(g);
      dispatch(p, [gh](int64_t i) { return f16_to_f32(gh[i]); }, m, v, p16t,
               n, lr, beta1, beta2, eps, weight_decay, bc1, bc2s, inv_scale,
               adamw != 0);
      break;
    }
  }
}

// ---- CPU Lion (reference csrc/lion/cpu_lion_impl.cpp): p -= lr*(sign(c)
// + wd*p), m = b2*m + (1-b2)*g, with c = b1*m + (1-b1)*g. Same fused
// streaming-pass design as Adam above.
namespace {

template <typename GradLoad, bool WRITE_BF16>
void lion_loop(float* __restrict__ p, GradLoad gload, float* __restrict__ m,
               uint16_t* __restrict__ p16, int64_t n, float lr, float beta1,
               float beta2, float wd, float inv_scale) {
#pragma omp parallel for schedule(static)
  for (int64_t c = 0; c < n; c += 4096) {
    const int64_t end = c + 4096 < n ? c + 4096 : n;
#pragma omp simd
    for (int64_t i = c; i < end; ++i) {
      const float g = gload(i) * inv_scale;
      const float u = beta1 * m[i] + (1.f - beta1) * g;
      const float s = u > 0.f ? 1.f : (u < 0.f ? -1.f : 0.f);
      p[i] -= lr * (s + wd * p[i]);
      m[i] = beta2 * m[i] + (1.f - beta2) * g;
      if (WRITE_BF16) p16[i] = f32_to_bf16_rne(p[i]);
    }
  }
}

template <typename GradLoad>
void lion_dispatch(float* p, GradLoad gload, float* m, uint16_t* p16,
                   int64_t n, float lr, float b1, float b2, float wd,
                   float inv_scale) {
  if (p16)
    lion_loop<GradLoad, true>(p, gload, m, p16, n, lr, b1, b2, wd, inv_scale);
  else
    lion_loop<GradLoad, false>(p, gload, m, p16, n, lr, b1, b2, wd,
                               inv_scale);
}

}  // namespace

extern "C" void ds_cpu_lion_flat(float* p, const void* g, int grad_dtype,
                                 float* m, void* p16, long long n, float lr,
                                 float beta1, float beta2, float weight_decay,
                                 float inv_scale) {
  uint16_t* p16t = reinterpret_cast<uint16_t*>(p16);
  switch (grad_dtype) {
    case 0: {
      const float* gf = reinterpret_cast<const float*>(g);
      lion_dispatch(p, [gf](int64_t i) { return gf[i]; }, m, p16t, n, lr,
                    beta1, beta2, weight_decay, inv_scale);
      break;
    }
    case 1: {
      const uint16_t* gb = reinterpret_cast<const uint16_t*>(g);
      lion_dispatch(p, [gb](int64_t i) { return bf16_to_f32(gb[i]); }, m,
                    p16t, n, lr, beta1, beta2, weight_decay, inv_scale);
      break;
    }
    case 2: {
      const uint16_t* gh = reinterpret_cast<const uint16_t*>(g);
      lion_dispatch(p, [gh](int64_t i) { return f16_to_f32(gh[i]); }, m, p16t,
                    n, lr, beta1, beta2, weight_decay, inv_scale);
      break;
    }
  }
}
