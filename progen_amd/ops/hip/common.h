// Common device helpers for ProGen CDNA4 (gfx950) kernels.
// Wavefront = 64; block sizes are multiples of 64 throughout.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

using bf16 = __hip_bfloat16;

// vector types for wide loads (16 B / lane sweet spot)
typedef short bf16x8 __attribute__((ext_vector_type(8)));   // 8 bf16 = 16 B
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

__device__ __forceinline__ float bf2f(short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f2bf(float f) {
  // __float2bfloat16 lowers to ONE v_cvt_pk_bf16_f32 (RNE, same
  // rounding as torch); the manual bit-math RNE this replaces was
  // three VALU ops in every kernel's store path
  __hip_bfloat16 b = __float2bfloat16(f);
  return *(short*)&b;
}

// wave-wide reductions (64 lanes)
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// reduce across a 16-lane subgroup (xor over lanes 1,2,4,8); used where the
// MFMA C-layout spreads one output row over lanes sharing (lane & 15)
__device__ __forceinline__ float group16_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float group16_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// fast tanh on v_exp_f32 (libm tanhf is a slow ocml polynomial — the
// round-2 profile put glu_bwd at 56% of HBM peak purely on VALU cost;
// v_exp is ~2 cycles beside other work). (e-1)/(e+1) with the argument
// clamped so exp never overflows to inf (inf/inf = NaN); |x| >= 15 is
// tanh = +-1 to beyond fp32 precision anyway.
__device__ __forceinline__ float fast_tanh(float x) {
  float xc = fminf(fmaxf(x, -15.0f), 15.0f);
  float e = __expf(2.0f * xc);
  return (e - 1.0f) / (e + 1.0f);
}

// tanh-approximation GELU (matches jax.nn.gelu approximate=True and
// torch F.gelu(approximate="tanh"))
__device__ __forceinline__ float gelu_tanh(float x) {
  const float k0 = 0.7978845608028654f;  // sqrt(2/pi)
  const float k1 = 0.044715f;
  float inner = k0 * (x + k1 * x * x * x);
  return 0.5f * x * (1.0f + fast_tanh(inner));
}

__device__ __forceinline__ float gelu_tanh_grad(float x) {
  const float k0 = 0.7978845608028654f;
  const float k1 = 0.044715f;
  float x2 = x * x;
  float inner = k0 * (x + k1 * x * x2);
  float t = fast_tanh(inner);
  float dinner = k0 * (1.0f + 3.0f * k1 * x2);
  return 0.5f * (1.0f + t) + 0.5f * x * (1.0f - t * t) * dinner;
}

// value AND derivative from ONE tanh (the backward needs both; two
// separate calls were the other half of glu_bwd's VALU bill)
__device__ __forceinline__ void gelu_tanh_both(float x, float* val,
                                               float* grad) {
  const float k0 = 0.7978845608028654f;
  const float k1 = 0.044715f;
  float x2 = x * x;
  float inner = k0 * (x + k1 * x * x2);
  float t = fast_tanh(inner);
  float dinner = k0 * (1.0f + 3.0f * k1 * x2);
  *val = 0.5f * x * (1.0f + t);
  *grad = 0.5f * (1.0f + t) + 0.5f * x * (1.0f - t * t) * dinner;
}

#define HIP_CHECK_LAST()                                                      \
  do {                                                                        \
    hipError_t e_ = hipGetLastError();                                        \
    if (e_ != hipSuccess) {                                                   \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e_), __FILE__,     \
             __LINE__);                                                       \
    }                                                                         \
  } while (0)
