// Semantic-chunking distance scorer: cosine distance between adjacent
// embedding rows. The reference computed this on CPU with a numpy loop
// (distllm/embed/embedders/semantic_chunk.py:24-55); here it is one HIP
// kernel over the device-resident embedding matrix.
//
// in:  emb [N, H] fp32 (pooled, possibly unnormalized)
// out: dist [N-1] fp32 ; dist[i] = 1 - dot(e_i, e_{i+1}) / (|e_i||e_{i+1}|)
#include "common.h"

__global__ void __launch_bounds__(256)
adjacent_cosine_kernel(const float* __restrict__ emb,
                       float* __restrict__ dist, int N, int H) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int waves = (gridDim.x * blockDim.x) >> 6;
  for (int i = blockIdx.x * 4 + wid; i < N - 1; i += waves) {
    const float* a = emb + (long)i * H;
    const float* b = a + H;
    float dot = 0.f, na = 0.f, nb = 0.f;
    for (int c = lane; c < H; c += WAVE) {
      float av = a[c], bv = b[c];
      dot += av * bv;
      na += av * av;
      nb += bv * bv;
    }
    dot = wave_reduce_sum(dot);
    na = wave_reduce_sum(na);
    nb = wave_reduce_sum(nb);
    if (lane == 0)
      dist[i] = 1.f - dot * rsqrtf(fmaxf(na, 1e-30f)) *
                          rsqrtf(fmaxf(nb, 1e-30f));
  }
}

extern "C" void launch_adjacent_cosine(const float* emb, float* dist, int N,
                                       int H, hipStream_t stream) {
  int blocks = min((N - 1 + 3) / 4, 2048);
  if (N < 2) return;
  hipLaunchKernelGGL(adjacent_cosine_kernel, dim3(blocks), dim3(256), 0,
                     stream, emb, dist, N, H);
}
