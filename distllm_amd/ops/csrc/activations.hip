// Fused elementwise activation kernels (bf16, vectorized s16x8 loads).
//
// bias_gelu: y = gelu(x + bias)   (BERT/ESM FFN; reference delegated to
//   HF transformers — SURVEY.md K1/K3 rows)
// swiglu:    y = silu(gate) * up  (Mistral/Llama FFN)
//
// Guide Appendix B: memory-bound; the only job is full-width vectorized
// HBM streaming. Grid-stride, <= 2048 blocks.
#include "common.h"

template <bool ERF>
__global__ void __launch_bounds__(256)
bias_gelu_kernel(const unsigned short* __restrict__ x,
                 const unsigned short* __restrict__ bias,
                 unsigned short* __restrict__ out, long n_elem, int N) {
  const long n_vec = n_elem >> 3;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < n_vec;
       v += stride) {
    s16x8 xv = *reinterpret_cast<const s16x8*>(x + v * 8);
    float f[8];
    bf8_to_f32(xv, f);
    if (bias != nullptr) {
      const int col = (int)((v * 8) % N);
      s16x8 bv = *reinterpret_cast<const s16x8*>(bias + col);
      float bf[8];
      bf8_to_f32(bv, bf);
#pragma unroll
      for (int i = 0; i < 8; ++i) f[i] += bf[i];
    }
    s16x8 ov;
#pragma unroll
    for (int i = 0; i < 8; ++i)
      ov[i] = (short)f2bf(ERF ? gelu_erf(f[i]) : fast_gelu(f[i]));
    *reinterpret_cast<s16x8*>(out + v * 8) = ov;
  }
}

__global__ void __launch_bounds__(256)
swiglu_kernel(const unsigned short* __restrict__ gate,
              const unsigned short* __restrict__ up,
              unsigned short* __restrict__ out, long n_elem) {
  const long n_vec = n_elem >> 3;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < n_vec;
       v += stride) {
    s16x8 gv = *reinterpret_cast<const s16x8*>(gate + v * 8);
    s16x8 uv = *reinterpret_cast<const s16x8*>(up + v * 8);
    float g[8], u[8];
    bf8_to_f32(gv, g);
    bf8_to_f32(uv, u);
    s16x8 ov;
#pragma unroll
    for (int i = 0; i < 8; ++i) ov[i] = (short)f2bf(silu(g[i]) * u[i]);
    *reinterpret_cast<s16x8*>(out + v * 8) = ov;
  }
}



// Packed swiglu: gu [R, 2I] (gate cols then up cols) -> out [R, I].
// Lets the fused gate|up GEMM output feed the MLP without slicing the
// halves into contiguous copies.
__global__ void __launch_bounds__(256)
swiglu_packed_kernel(const unsigned short* __restrict__ gu,
                     unsigned short* __restrict__ out, long R, int I) {
  const long n_vec = (R * (long)I) >> 3;
  const long stride = (long)gridDim.x * blockDim.x;
  const int iv = I >> 3;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < n_vec;
       v += stride) {
    const long r = v / iv;
    const int c = (int)(v % iv);
    const unsigned short* row = gu + r * (2L * I);
    s16x8 gv = *reinterpret_cast<const s16x8*>(row + c * 8);
    s16x8 uv = *reinterpret_cast<const s16x8*>(row + I + c * 8);
    float g[8], u[8];
    bf8_to_f32(gv, g);
    bf8_to_f32(uv, u);
    s16x8 ov;
#pragma unroll
    for (int i = 0; i < 8; ++i) ov[i] = (short)f2bf(silu(g[i]) * u[i]);
    *reinterpret_cast<s16x8*>(out + v * 8) = ov;
  }
}

extern "C" void launch_swiglu_packed(const void* gu, void* out, long R,
                                     int I, hipStream_t stream) {
  const long n_vec = (R * (long)I) >> 3;
  int blocks = (int)min((n_vec + 255) / 256, (long)2048);
  hipLaunchKernelGGL(swiglu_packed_kernel, dim3(blocks), dim3(256), 0,
                     stream, (const unsigned short*)gu,
                     (unsigned short*)out, R, I);
}

// Fused bias-add (no activation), used for attention output projections
// where hipBLASLt's bias epilogue is not engaged.
__global__ void __launch_bounds__(256)
bias_add_kernel(const unsigned short* __restrict__ x,
                const unsigned short* __restrict__ bias,
                unsigned short* __restrict__ out, long n_elem, int N) {
  const long n_vec = n_elem >> 3;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < n_vec;
       v += stride) {
    s16x8 xv = *reinterpret_cast<const s16x8*>(x + v * 8);
    const int col = (int)((v * 8) % N);
    s16x8 bv = *reinterpret_cast<const s16x8*>(bias + col);
    float f[8], bf[8];
    bf8_to_f32(xv, f);
    bf8_to_f32(bv, bf);
    s16x8 ov;
#pragma unroll
    for (int i = 0; i < 8; ++i) ov[i] = (short)f2bf(f[i] + bf[i]);
    *reinterpret_cast<s16x8*>(out + v * 8) = ov;
  }
}

static int grid_for(long n_elem) {
  long n_vec = n_elem >> 3;
  long b = (n_vec + 255) / 256;
  return (int)min(b, (long)2048);
}

extern "C" void launch_bias_gelu(const void* x, const void* bias, void* out,
                                 long n_elem, int N, bool erf,
                                 hipStream_t stream) {
  if (erf)
    hipLaunchKernelGGL((bias_gelu_kernel<true>), dim3(grid_for(n_elem)),
                       dim3(256), 0, stream, (const unsigned short*)x,
                       (const unsigned short*)bias, (unsigned short*)out,
                       n_elem, N);
  else
    hipLaunchKernelGGL((bias_gelu_kernel<false>), dim3(grid_for(n_elem)),
                       dim3(256), 0, stream, (const unsigned short*)x,
                       (const unsigned short*)bias, (unsigned short*)out,
                       n_elem, N);
}

extern "C" void launch_swiglu(const void* gate, const void* up, void* out,
                              long n_elem, hipStream_t stream) {
  hipLaunchKernelGGL(swiglu_kernel, dim3(grid_for(n_elem)), dim3(256), 0,
                     stream, (const unsigned short*)gate,
                     (const unsigned short*)up, (unsigned short*)out, n_elem);
}

extern "C" void launch_bias_add(const void* x, const void* bias, void* out,
                                long n_elem, int N, hipStream_t stream) {
  hipLaunchKernelGGL(bias_add_kernel, dim3(grid_for(n_elem)), dim3(256), 0,
                     stream, (const unsigned short*)x,
                     (const unsigned short*)bias, (unsigned short*)out, n_elem,
                     N);
}
