// RoPE (rotary position embedding) application, half-split (NeoX) style:
//   out[..., :D/2] = x1*cos - x2*sin ; out[..., D/2:] = x2*cos + x1*sin
// used by Mistral/Llama and ESM2 (K3).
//
// cos/sin tables are HOST-precomputed f32 [S, D/2] (guide Appendix B:
// on-device trig turns memory-bound into VALU-bound).
// x: [B, S, Hx, D] bf16, in-place or out-of-place; pos gives the
// absolute position of each token row (int32 [B, S]) so the same kernel
// serves prefill (pos = arange) and decode (pos = context offsets).
#include "common.h"

__global__ void __launch_bounds__(256)
rope_kernel(const unsigned short* __restrict__ x,
            unsigned short* __restrict__ out,
            const float* __restrict__ cs,   // [max_pos, D/2] cos
            const float* __restrict__ sn,   // [max_pos, D/2] sin
            const int* __restrict__ pos,    // [B*S] absolute positions
            long n_rows,                    // B*S*Hx
            int Hx, int D, int rope_heads) {
  // heads >= rope_heads pass through unchanged: lets ONE launch rope
  // the q|k part of a fused qkv tensor while copying v (decode-path
  // launch fusion)
  const int half = D / 2;
  const long stride = (long)gridDim.x * blockDim.x;
  const long total = n_rows * half;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const long row = i / half;
    const int d = (int)(i % half);
    const long tok = row / Hx;
    if ((int)(row - tok * Hx) >= rope_heads) {
      const unsigned short* xr = x + row * D;
      out[row * D + d] = xr[d];
      out[row * D + d + half] = xr[d + half];
      continue;
    }
    const int p = pos[tok];
    const float c = cs[(long)p * half + d];
    const float s = sn[(long)p * half + d];
    const unsigned short* xr = x + row * D;
    float x1 = bf2f(xr[d]);
    float x2 = bf2f(xr[d + half]);
    out[row * D + d] = f2bf(x1 * c - x2 * s);
    out[row * D + d + half] = f2bf(x2 * c + x1 * s);
  }
}

extern "C" void launch_rope(const void* x, void* out, const float* cs,
                            const float* sn, const int* pos, long n_rows,
                            int Hx, int D, int rope_heads,
                            hipStream_t stream) {
  const long total = n_rows * (D / 2);
  int blocks = (int)min((total + 255) / 256, (long)2048);
  hipLaunchKernelGGL(rope_kernel, dim3(blocks), dim3(256), 0, stream,
                     (const unsigned short*)x, (unsigned short*)out, cs, sn,
                     pos, n_rows, Hx, D, rope_heads);
}
