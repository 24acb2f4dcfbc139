// kNN search kernels replacing faiss-gpu (reference rag/search.py:231-399,
// SURVEY.md K11/K12/K14): chunked radix-select top-k over a score matrix,
// and packed-binary Hamming scoring for the ubinary precision path.
//
// Pipeline (python side, rag/search.py here):
//   scores_chunk = Q @ C_chunk^T          (hipBLASLt or gemm.hip)
//   topk_chunk_kernel -> per-chunk candidates
//   topk over concatenated candidates (same kernel, with idx indirection)
//
// topk_chunk_kernel: one 256-thread block per (query, chunk). Two-round
// 8-bit radix select over monotonic uint32 keys finds a threshold, a
// compact pass gathers candidates (cap 2048) into LDS, a bitonic sort
// orders them, and the top k (vals + indices) are written out.
#include "common.h"

#define TK_CAP 2048  // candidate buffer (vals+idx) in LDS

__device__ __forceinline__ unsigned int f32_to_ord(float f) {
  union { float f; unsigned int u; } c;
  c.f = f;
  return (c.u & 0x80000000u) ? ~c.u : (c.u | 0x80000000u);
}
__device__ __forceinline__ float ord_to_f32(unsigned int u) {
  union { float f; unsigned int u; } c;
  c.u = (u & 0x80000000u) ? (u & 0x7fffffffu) : ~u;
  return c.f;
}

// scores: [Q, N] f32. in_idx: optional [Q, N] int32 (indirection from a
// previous candidate round); out: [Q, nchunks, k] vals f32 + idx int32.
__global__ void __launch_bounds__(256)
topk_chunk_kernel(const float* __restrict__ scores,
                  const int* __restrict__ in_idx,
                  float* __restrict__ out_v, int* __restrict__ out_i,
                  long N, int chunk, int k, int nchunks) {
  __shared__ unsigned int hist[256];
  __shared__ unsigned int cnt_hi, thresh, ncand;
  __shared__ unsigned int cand_u[TK_CAP];
  __shared__ int cand_i[TK_CAP];

  const int q = blockIdx.y;
  const int c = blockIdx.x;
  const long c0 = (long)c * chunk;
  const long c1 = min(c0 + chunk, N);
  const float* row = scores + (long)q * N;

  // ---- round 1: top-8-bit histogram ----------------------------------
  if (threadIdx.x < 256) hist[threadIdx.x] = 0;
  __syncthreads();
  for (long i = c0 + threadIdx.x; i < c1; i += 256)
    atomicAdd(&hist[f32_to_ord(row[i]) >> 24], 1u);
  __syncthreads();
  // scan bins from high to low on thread 0 (256 steps, trivial)
  if (threadIdx.x == 0) {
    unsigned int acc = 0;
    int b1 = 0;
    for (int b = 255; b >= 0; --b) {
      if (acc + hist[b] >= (unsigned int)k || b == 0) { b1 = b; break; }
      acc += hist[b];
    }
    cnt_hi = acc;
    thresh = (unsigned int)b1 << 24;
  }
  __syncthreads();
  const unsigned int b1 = thresh >> 24;
  const unsigned int need2 = (unsigned int)k - cnt_hi;

  // ---- round 2: next-8-bit histogram within bin b1 --------------------
  if (threadIdx.x < 256) hist[threadIdx.x] = 0;
  __syncthreads();
  for (long i = c0 + threadIdx.x; i < c1; i += 256) {
    unsigned int u = f32_to_ord(row[i]);
    if ((u >> 24) == b1) atomicAdd(&hist[(u >> 16) & 0xffu], 1u);
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned int acc = 0;
    int b2 = 0;
    for (int b = 255; b >= 0; --b) {
      if (acc + hist[b] >= need2 || b == 0) { b2 = b; break; }
      acc += hist[b];
    }
    thresh = (b1 << 24) | ((unsigned int)b2 << 16);
    ncand = 0;
  }
  __syncthreads();
  const unsigned int T = thresh;

  // ---- compact pass: gather everything with u >= T --------------------
  for (long i = c0 + threadIdx.x; i < c1; i += 256) {
    unsigned int u = f32_to_ord(row[i]);
    if (u >= T) {
      unsigned int slot = atomicAdd(&ncand, 1u);
      if (slot < TK_CAP) {
        cand_u[slot] = u;
        cand_i[slot] = in_idx ? in_idx[(long)q * N + i] : (int)i;
      }
    }
  }
  __syncthreads();
  int n = (int)min(ncand, (unsigned int)TK_CAP);
  // pad to power of two (covering k) for bitonic
  const int target = max(n, k);
  int n2 = 1;
  while (n2 < target) n2 <<= 1;
  for (int i = n + threadIdx.x; i < n2; i += 256) {
    cand_u[i] = 0;       // smallest ordinal
    cand_i[i] = -1;
  }
  __syncthreads();

  // ---- bitonic sort descending over n2 elements -----------------------
  for (int size = 2; size <= n2; size <<= 1) {
    for (int stride = size >> 1; stride > 0; stride >>= 1) {
      for (int i = threadIdx.x; i < n2 / 2; i += 256) {
        const int lo = 2 * i - (i & (stride - 1));
        const int hi = lo + stride;
        const bool desc = ((lo & size) == 0);
        unsigned int a = cand_u[lo], b = cand_u[hi];
        if ((a < b) == desc) {
          int ai = cand_i[lo], bi = cand_i[hi];
          cand_u[lo] = b; cand_u[hi] = a;
          cand_i[lo] = bi; cand_i[hi] = ai;
        }
      }
      __syncthreads();
    }
  }

  // ---- write top-k ----------------------------------------------------
  float* ov = out_v + ((long)q * nchunks + c) * k;
  int* oi = out_i + ((long)q * nchunks + c) * k;
  for (int i = threadIdx.x; i < k; i += 256) {
    if (i < n) {
      ov[i] = ord_to_f32(cand_u[i]);
      oi[i] = cand_i[i];
    } else {
      ov[i] = -INFINITY;
      oi[i] = -1;
    }
  }
}

extern "C" void launch_topk_chunk(const float* scores, const int* in_idx,
                                  float* out_v, int* out_i, int Qn, long N,
                                  int chunk, int k, hipStream_t stream) {
  const int nchunks = (int)((N + chunk - 1) / chunk);
  dim3 grid(nchunks, Qn);
  hipLaunchKernelGGL(topk_chunk_kernel, grid, dim3(256), 0, stream, scores,
                     in_idx, out_v, out_i, N, chunk, k, nchunks);
}

// ---- Hamming distance scoring (ubinary precision path) -----------------
// codes are WORD-MAJOR: codes_t [W, N] u64 (W = nbits/64) so candidate
// reads coalesce; query: [Q, W] u64. out: [Q, N] f32 = -hamming.
__global__ void __launch_bounds__(256)
hamming_score_kernel(const unsigned long long* __restrict__ qcodes,
                     const unsigned long long* __restrict__ codes_t,
                     float* __restrict__ out, long N, int W) {
  const int q = blockIdx.y;
  const long stride = (long)gridDim.x * blockDim.x;
  // query words via L2/L1 (small)
  for (long n = (long)blockIdx.x * blockDim.x + threadIdx.x; n < N;
       n += stride) {
    int d = 0;
    for (int w = 0; w < W; ++w)
      d += __popcll(qcodes[(long)q * W + w] ^ codes_t[(long)w * N + n]);
    out[(long)q * N + n] = -(float)d;
  }
}

extern "C" void launch_hamming_score(const void* qcodes, const void* codes_t,
                                     float* out, int Qn, long N, int W,
                                     hipStream_t stream) {
  int bl = (int)min((N + 255) / 256, (long)2048);
  dim3 grid(bl, Qn);
  hipLaunchKernelGGL(hamming_score_kernel, grid, dim3(256), 0, stream,
                     (const unsigned long long*)qcodes,
                     (const unsigned long long*)codes_t, out, N, W);
}

// ---- batched IVF chunk scoring (approximate search) --------
// slabs: [nchunks, cap, H] f32 padded chunk slabs; probes: [Q*nprobe]
// int32 chunk ids; qv: [Q, H] f32; out: [Q*nprobe*cap] f32 (-inf at
// pad rows). One block per (query, probe): the query vector is staged
// in LDS once; each wave streams rows of the contiguous slab with
// 16-byte-per-lane coalesced loads and a wave-reduce dot. BW-bound by
// design (bf16 slabs read once; winners re-scored fp32 host-side).
__global__ void __launch_bounds__(256)
ivf_score_kernel(const float* __restrict__ slabs,
                 const int* __restrict__ lens,
                 const int* __restrict__ probes,
                 const float* __restrict__ qv, float* __restrict__ out,
                 int nprobe, int cap, int H) {
  extern __shared__ __attribute__((aligned(16))) float qlds[];
  const int qi = blockIdx.x / nprobe;
  const int ch = probes[blockIdx.x];
  const int len = lens[ch];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;

  for (int i = threadIdx.x; i < H; i += 256) qlds[i] = qv[qi * H + i];
  __syncthreads();

  const float* slab = slabs + (long)ch * cap * H;
  float* orow = out + (long)blockIdx.x * cap;
  const int span = 64 * 4;  // elements covered per wave pass
  for (int r = wid; r < cap; r += 4) {
    float acc = 0.f;
    if (r < len) {
      const float* row = slab + (long)r * H;
      for (int i0 = lane * 4; i0 < H; i0 += span) {
        // 4 f32 = one 16-byte load (tail guard per element)
        if (i0 + 4 <= H) {
          const f32x4 v = *reinterpret_cast<const f32x4*>(row + i0);
#pragma unroll
          for (int j = 0; j < 4; ++j) acc += v[j] * qlds[i0 + j];
        } else {
          for (int j = i0; j < H; ++j) acc += row[j] * qlds[j];
        }
      }
      acc = wave_reduce_sum(acc);
    }
    if (lane == 0) orow[r] = (r < len) ? acc : -INFINITY;
  }
}

extern "C" void launch_ivf_score(const void* slabs, const int* lens,
                                 const int* probes, const float* qv,
                                 float* out, int nblocks, int nprobe,
                                 int cap, int H, hipStream_t stream) {
  hipLaunchKernelGGL(ivf_score_kernel, dim3(nblocks), dim3(256),
                     H * (int)sizeof(float), stream,
                     (const float*)slabs, lens, probes, qv, out,
                     nprobe, cap, H);
}
