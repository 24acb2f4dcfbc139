// Common device helpers for distllm-amd CDNA4 (gfx950) kernels.
// Hardware model per /opt/skills/guides/cdna_hip_programming.md:
//   wave = 64 lanes, 4 SIMD-32/CU, 160 KiB LDS/CU, MFMA bf16 K=16/32.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64

// ---- vector types -------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) float  f32x2;
typedef __attribute__((ext_vector_type(4))) float  f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) short  s16x4;
typedef __attribute__((ext_vector_type(8))) short  s16x8;
typedef __attribute__((ext_vector_type(2))) int    i32x2;
typedef __attribute__((ext_vector_type(4))) int    i32x4;

// bf16 <-> f32 (scalar); use __hip_bfloat16 for storage only.
__device__ __forceinline__ float bf2f(unsigned short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}
__device__ __forceinline__ unsigned short f2bf(float f) {
  // hardware RTNE converter (v_cvt_*_bf16_f32 on gfx950) — far cheaper
  // than the 5-op manual rounding sequence.
  __bf16 h = (__bf16)f;
  union { __bf16 h; unsigned short u; } c;
  c.h = h;
  return c.u;
}

// Convert 8 bf16 (s16x8) to 8 f32.
__device__ __forceinline__ void bf8_to_f32(const s16x8 v, float* out) {
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = bf2f((unsigned short)v[i]);
}

// ---- wave reductions (64-wide) -----------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}
__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block-level reduce across NW waves using LDS scratch (NW <= 16).
template <int NW>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float r = (lane < NW) ? lds_scratch[lane] : 0.f;
  r = wave_reduce_sum(r);
  // broadcast via readfirstlane semantics: all lanes of wave 0 now agree
  if (threadIdx.x == 0) lds_scratch[0] = r;
  __syncthreads();
  float out = lds_scratch[0];
  __syncthreads();
  return out;
}

#define HIP_CHECK_KERNEL()                                                    \
  do {                                                                        \
    hipError_t e = hipGetLastError();                                         \
    if (e != hipSuccess) {                                                    \
      printf("HIP kernel launch error: %s\n", hipGetErrorString(e));          \
    }                                                                         \
  } while (0)

// Branch-free transcendentals on the native v_exp_f32 / v_rcp_f32
// instructions. libm tanhf/erff/expf compile to branchy range-checked
// slow paths (one s_and_saveexec waterfall per element — measured 37%
// of a fused-GELU GEMM's time); these stay straight-line VALU.
__device__ __forceinline__ float fast_exp(float x) {
  return __builtin_amdgcn_exp2f(x * 1.4426950408889634f);
}
__device__ __forceinline__ float fast_rcp(float x) {
  return __builtin_amdgcn_rcpf(x);  // ~1 ulp: plenty for bf16 outputs
}
__device__ __forceinline__ float fast_tanh(float u) {
  // 1 - 2/(e^{2u}+1); exp2 under/overflow saturates to +-1 gracefully
  const float e = __builtin_amdgcn_exp2f(u * 2.885390081777927f);
  return 1.f - 2.f * fast_rcp(e + 1.f);
}
__device__ __forceinline__ float fast_erf(float z) {
  // Abramowitz-Stegun 7.1.26 (max abs err 1.5e-7), branch-free
  const float a = fabsf(z);
  const float t = fast_rcp(fmaf(0.3275911f, a, 1.f));
  float p = fmaf(1.061405429f, t, -1.453152027f);
  p = fmaf(p, t, 1.421413741f);
  p = fmaf(p, t, -0.284496736f);
  p = fmaf(p, t, 0.254829592f);
  const float y = 1.f - p * t * fast_exp(-a * a);
  return copysignf(y, z);
}
__device__ __forceinline__ float fast_gelu(float x) {
  // tanh approximation (matches BERT/HF "gelu_new" semantics)
  const float k0 = 0.7978845608028654f;  // sqrt(2/pi)
  const float k1 = 0.044715f;
  return 0.5f * x * (1.f + fast_tanh(k0 * (x + k1 * x * x * x)));
}
__device__ __forceinline__ float gelu_erf(float x) {
  return 0.5f * x * (1.f + fast_erf(x * 0.70710678118654752f));
}
__device__ __forceinline__ float silu(float x) {
  return x * fast_rcp(1.f + fast_exp(-x));
}
