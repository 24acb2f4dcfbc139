// Pooling kernels: masked mean (excl. BOS/EOS) and last-token gather,
// each with fused L2 normalization; fp32 accumulate, fp32 output.
//
// Semantics parity with reference distllm/embed/poolers/mean.py:13-49
// (mask[:,0] and mask[:,L-1] zeroed, clamped count) and
// last_token.py:12-38 (gather at L-1, or S-1 when left-padded), plus the
// F.normalize(p=2) of embedders/full_sequence.py:69 fused in.
#include "common.h"

// hidden [R, S, H] bf16, mask [R, S] int32 -> out [R, H] fp32
template <int NCT>  // per-thread vec8 column chunks (ceil(H/8/256))
__global__ void __launch_bounds__(256)
mean_pool_kernel(const unsigned short* __restrict__ hidden,
                 const int* __restrict__ mask, float* __restrict__ out,
                 int R, int S, int H, bool normalize) {
  __shared__ float lds[16];
  const int n_chunks = H >> 3;

  for (int r = blockIdx.x; r < R; r += gridDim.x) {
    // 1) row length L = sum(mask[r,:])
    int lsum = 0;
    for (int s = threadIdx.x; s < S; s += blockDim.x) lsum += mask[r * S + s];
    float L = block_reduce_sum<4>((float)lsum, lds);
    const int Li = (int)(L + 0.5f);
    const float count = fmaxf((float)Li - 2.f, 1.f);

    // 2) accumulate sum over s in [1, Li-2] (BOS=0 and last=Li-1 excluded)
    float acc[NCT][8];
#pragma unroll
    for (int k = 0; k < NCT; ++k)
#pragma unroll
      for (int i = 0; i < 8; ++i) acc[k][i] = 0.f;

    for (int s = 1; s < Li - 1; ++s) {
      if (s >= S) break;
      if (!mask[r * S + s]) continue;  // non-contiguous masks still honored
      const unsigned short* hrow = hidden + ((long)r * S + s) * H;
#pragma unroll
      for (int k = 0; k < NCT; ++k) {
        const int c = threadIdx.x + k * 256;
        if (c < n_chunks) {
          s16x8 hv = *reinterpret_cast<const s16x8*>(hrow + c * 8);
          float f[8];
          bf8_to_f32(hv, f);
#pragma unroll
          for (int i = 0; i < 8; ++i) acc[k][i] += f[i];
        }
      }
    }

    // 3) divide; optional fused L2 norm; write fp32
    float sq = 0.f;
#pragma unroll
    for (int k = 0; k < NCT; ++k) {
      const int c = threadIdx.x + k * 256;
      if (c < n_chunks) {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          acc[k][i] /= count;
          sq += acc[k][i] * acc[k][i];
        }
      }
    }
    float scale = 1.f;
    if (normalize) {
      __syncthreads();
      float tot = block_reduce_sum<4>(sq, lds);
      scale = rsqrtf(fmaxf(tot, 1e-24f));
    }
#pragma unroll
    for (int k = 0; k < NCT; ++k) {
      const int c = threadIdx.x + k * 256;
      if (c < n_chunks) {
        f32x4 lo, hi;
#pragma unroll
        for (int i = 0; i < 4; ++i) lo[i] = acc[k][i] * scale;
#pragma unroll
        for (int i = 0; i < 4; ++i) hi[i] = acc[k][4 + i] * scale;
        *reinterpret_cast<f32x4*>(out + (long)r * H + c * 8) = lo;
        *reinterpret_cast<f32x4*>(out + (long)r * H + c * 8 + 4) = hi;
      }
    }
    __syncthreads();
  }
}

// last-token gather: out[r,:] = hidden[r, idx, :]; idx = L-1 (right pad)
// or S-1 (left pad); fused L2 normalize.
__global__ void __launch_bounds__(256)
last_token_pool_kernel(const unsigned short* __restrict__ hidden,
                       const int* __restrict__ mask, float* __restrict__ out,
                       int R, int S, int H, bool left_padded, bool normalize) {
  __shared__ float lds[16];
  const int n_chunks = H >> 3;
  for (int r = blockIdx.x; r < R; r += gridDim.x) {
    int idx;
    if (left_padded) {
      idx = S - 1;
    } else {
      int lsum = 0;
      for (int s = threadIdx.x; s < S; s += blockDim.x)
        lsum += mask[r * S + s];
      float L = block_reduce_sum<4>((float)lsum, lds);
      idx = max((int)(L + 0.5f) - 1, 0);
    }
    const unsigned short* hrow = hidden + ((long)r * S + idx) * H;
    float sq = 0.f;
    float vals[8][8];  // up to H=16384
    int nk = (n_chunks + 255) / 256;
#pragma unroll 8
    for (int k = 0; k < nk && k < 8; ++k) {
      const int c = threadIdx.x + k * 256;
      if (c < n_chunks) {
        s16x8 hv = *reinterpret_cast<const s16x8*>(hrow + c * 8);
        bf8_to_f32(hv, vals[k]);
#pragma unroll
        for (int i = 0; i < 8; ++i) sq += vals[k][i] * vals[k][i];
      }
    }
    float scale = 1.f;
    if (normalize) {
      __syncthreads();
      float tot = block_reduce_sum<4>(sq, lds);
      scale = rsqrtf(fmaxf(tot, 1e-24f));
    }
#pragma unroll 8
    for (int k = 0; k < nk && k < 8; ++k) {
      const int c = threadIdx.x + k * 256;
      if (c < n_chunks) {
#pragma unroll
        for (int i = 0; i < 8; ++i)
          out[(long)r * H + c * 8 + i] = vals[k][i] * scale;
      }
    }
    __syncthreads();
  }
}

// standalone row-wise L2 normalize, fp32 in/out [R, H]
__global__ void __launch_bounds__(256)
l2_normalize_kernel(const float* __restrict__ x, float* __restrict__ out,
                    int R, int H) {
  __shared__ float lds[16];
  for (int r = blockIdx.x; r < R; r += gridDim.x) {
    float sq = 0.f;
    for (int c = threadIdx.x; c < H; c += blockDim.x) {
      float v = x[(long)r * H + c];
      sq += v * v;
    }
    float tot = block_reduce_sum<4>(sq, lds);
    float scale = rsqrtf(fmaxf(tot, 1e-24f));
    for (int c = threadIdx.x; c < H; c += blockDim.x)
      out[(long)r * H + c] = x[(long)r * H + c] * scale;
    __syncthreads();
  }
}

extern "C" void launch_mean_pool(const void* hidden, const int* mask,
                                 float* out, int R, int S, int H,
                                 bool normalize, hipStream_t stream) {
  int blocks = min(R, 2048);
  int nct = (H / 8 + 255) / 256;
#define DM(N)                                                                  \
  hipLaunchKernelGGL((mean_pool_kernel<N>), dim3(blocks), dim3(256), 0,        \
                     stream, (const unsigned short*)hidden, mask, out, R, S,   \
                     H, normalize)
  switch (nct) {
    case 1: DM(1); break;
    case 2: DM(2); break;
    case 3: DM(3); break;
    case 4: DM(4); break;
    default: DM(8); break;
  }
#undef DM
}

extern "C" void launch_last_token_pool(const void* hidden, const int* mask,
                                       float* out, int R, int S, int H,
                                       bool left_padded, bool normalize,
                                       hipStream_t stream) {
  hipLaunchKernelGGL(last_token_pool_kernel, dim3(min(R, 2048)), dim3(256), 0,
                     stream, (const unsigned short*)hidden, mask, out, R, S, H,
                     left_padded, normalize);
}

extern "C" void launch_l2_normalize(const float* x, float* out, int R, int H,
                                    hipStream_t stream) {
  hipLaunchKernelGGL(l2_normalize_kernel, dim3(min(R, 2048)), dim3(256), 0,
                     stream, x, out, R, H);
}
