// Fused sampling kernel for the generation engine:
// temperature + min-p filter + categorical draw in one pass family,
// plus greedy argmax. Replaces vLLM's sampler (reference
// generate/generators/vllm_backend.py:46-60 defaults: temp=.5, min_p=.1).
//
// logits: [B, V] f32 ; rand: [B] uniform f32 ; out: [B] int32.
// min-p filter in logit space: keep l_i >= max + T*ln(min_p).
// Deterministic partition: thread t owns strided elements t, t+256, ...;
// per-thread partials are exclusive-scanned so the CDF walk is exact.
#include "common.h"

__global__ void __launch_bounds__(256)
sample_kernel(const float* __restrict__ logits,
              const float* __restrict__ rand, int* __restrict__ out, int V,
              float temp, float min_p) {
  __shared__ float lds[16];
  __shared__ float psum[257];
  const int b = blockIdx.x;
  const float* row = logits + (long)b * V;

  // pass 1: max
  float mx = -INFINITY;
  for (int i = threadIdx.x; i < V; i += 256) mx = fmaxf(mx, row[i]);
  {
    const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
    float w = wave_reduce_max(mx);
    if (lane == 0) lds[wid] = w;
    __syncthreads();
    float r = (lane < 4) ? lds[lane] : -INFINITY;
    r = wave_reduce_max(r);
    if (threadIdx.x == 0) lds[8] = r;
    __syncthreads();
    mx = lds[8];
  }

  const float invT = 1.f / temp;
  const float cut = (min_p > 0.f) ? mx + temp * __logf(min_p) : -INFINITY;

  // pass 2: per-thread sum of filtered exp
  float s = 0.f;
  for (int i = threadIdx.x; i < V; i += 256) {
    float l = row[i];
    if (l >= cut) s += fast_exp((l - mx) * invT);
  }
  psum[threadIdx.x] = s;
  __syncthreads();
  // exclusive scan over 256 partials (simple Hillis-Steele in LDS)
  for (int off = 1; off < 256; off <<= 1) {
    float v = psum[threadIdx.x];
    float add = (threadIdx.x >= off) ? psum[threadIdx.x - off] : 0.f;
    __syncthreads();
    psum[threadIdx.x] = v + add;
    __syncthreads();
  }
  const float total = psum[255];
  const float base = (threadIdx.x == 0) ? 0.f : psum[threadIdx.x - 1];
  const float target = rand[b] * total;

  // pass 3: the owning thread walks its strided chunk
  if (threadIdx.x == 0) out[b] = -1;
  __syncthreads();
  if (target >= base && target < base + s && s > 0.f) {
    float acc = base;
    for (int i = threadIdx.x; i < V; i += 256) {
      float l = row[i];
      if (l >= cut) {
        acc += fast_exp((l - mx) * invT);
        if (acc > target) { out[b] = i; break; }
      }
    }
  }
  __syncthreads();
  // numerical-edge fallback: pick argmax if nothing was selected
  if (threadIdx.x == 0 && out[b] < 0) {
    // rare: target == total due to rounding; fall back to max logit
    float bm = -INFINITY;
    int bi = 0;
    for (int i = 0; i < V; ++i)
      if (row[i] > bm) { bm = row[i]; bi = i; }
    out[b] = bi;
  }
}

__global__ void __launch_bounds__(256)
argmax_kernel(const float* __restrict__ logits, int* __restrict__ out,
              int V) {
  __shared__ float lv[4];
  __shared__ int li[4];
  const int b = blockIdx.x;
  const float* row = logits + (long)b * V;
  float mx = -INFINITY;
  int mi = 0;
  for (int i = threadIdx.x; i < V; i += 256) {
    if (row[i] > mx) { mx = row[i]; mi = i; }
  }
  // wave reduce argmax
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float om = __shfl_xor(mx, off, 64);
    int oi = __shfl_xor(mi, off, 64);
    if (om > mx || (om == mx && oi < mi)) { mx = om; mi = oi; }
  }
  if (lane == 0) { lv[wid] = mx; li[wid] = mi; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float bm = lv[0];
    int bi = li[0];
    for (int w = 1; w < 4; ++w)
      if (lv[w] > bm || (lv[w] == bm && li[w] < bi)) { bm = lv[w]; bi = li[w]; }
    out[b] = bi;
  }
}

extern "C" void launch_sample(const float* logits, const float* rand,
                              int* out, int B, int V, float temp, float min_p,
                              hipStream_t stream) {
  if (temp <= 0.f) {
    hipLaunchKernelGGL(argmax_kernel, dim3(B), dim3(256), 0, stream, logits,
                       out, V);
  } else {
    hipLaunchKernelGGL(sample_kernel, dim3(B), dim3(256), 0, stream, logits,
                       rand, out, V, temp, min_p);
  }
}
