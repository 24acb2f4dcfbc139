// Paged-KV decode attention (single new token per sequence), bf16.
// Replaces vLLM's paged attention (reference delegated generation to
// vLLM, generate/generators/vllm_backend.py).
//
// Cache layout: kcache/vcache [nblocks, HKV, BLK, D] bf16, BLK=16.
// block_table [B, max_blocks] int32 ; ctx_len [B] int32.
// q: [B, H, D] bf16 ; out: [B, H, D] bf16.
//
// Decode is latency-bound at small contexts: the v2 kernel stages the
// block table in LDS once, keeps the QK dot lane-per-key (independent
// vector loads), and runs the PV accumulation branch-free with an
// 8-deep unroll so the per-key V reads pipeline instead of serializing
// on L2 latency. Cross-wave combine via LDS.
#include "common.h"
#include <hip/hip_fp8.h>

#define PBLK 16  // tokens per KV block

__device__ __forceinline__ float fp8_to_f32(unsigned char u) {
  __hip_fp8_e4m3 v;
  v.__x = u;
  return float(v);
}

// CT: cache element type (unsigned short = bf16, unsigned char = OCP
// fp8 e4m3 — CDNA4-native; halves decode KV bandwidth, doubles
// capacity; values are stored unscaled, |k|,|v| << 448 in practice).
template <int D, typename CT>
__global__ void __launch_bounds__(256)
paged_attn_kernel(const unsigned short* __restrict__ q,
                  const CT* __restrict__ kcache,
                  const CT* __restrict__ vcache,
                  const int* __restrict__ block_table,
                  const int* __restrict__ ctx_len,
                  unsigned short* __restrict__ out, int B, int H, int HKV,
                  int max_blocks, float scale) {
  const int b = blockIdx.x / H;
  const int h = blockIdx.x % H;
  const int kvh = h / (H / HKV);
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int L = ctx_len[b];

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* p_lds = reinterpret_cast<float*>(smem);                 // 4*64
  float* q_lds = p_lds + 4 * 64;                                 // D
  float* wstate = q_lds + D;                                     // 8
  float* wacc = wstate + 8;                                      // 4*D
  int* bt_lds = reinterpret_cast<int*>(wacc + 4 * D);            // nblk

  const int nblk = (L + PBLK - 1) / PBLK;
  for (int i = threadIdx.x; i < nblk; i += 256)
    bt_lds[i] = block_table[(long)b * max_blocks + i];
  for (int d = threadIdx.x; d < D; d += 256)
    q_lds[d] = bf2f(q[((long)b * H + h) * D + d]) * scale;
  __syncthreads();

  float m = -1e30f, l = 0.f;
  float acc[2] = {0.f, 0.f};
  constexpr int DPL = D / 64;  // d-cols per lane (1 or 2)
  const long blk_stride = (long)HKV * PBLK * D;
  const long head_off = (long)kvh * PBLK * D;

  for (int t0 = wid * 64; t0 < L; t0 += 256) {
    const int key = t0 + lane;
    float s = -INFINITY;
    if (key < L) {
      const CT* krow = kcache +
          (long)bt_lds[key / PBLK] * blk_stride + head_off +
          (key % PBLK) * D;
      float dot = 0.f;
#pragma unroll
      for (int c = 0; c < D / 8; ++c) {
        float kf[8];
        if constexpr (sizeof(CT) == 2) {
          s16x8 kv8 = *reinterpret_cast<const s16x8*>(krow + c * 8);
          bf8_to_f32(kv8, kf);
        } else {
          i32x2 kv8 = *reinterpret_cast<const i32x2*>(krow + c * 8);
          const unsigned char* b =
              reinterpret_cast<const unsigned char*>(&kv8);
#pragma unroll
          for (int i = 0; i < 8; ++i) kf[i] = fp8_to_f32(b[i]);
        }
#pragma unroll
        for (int i = 0; i < 8; ++i) dot += kf[i] * q_lds[c * 8 + i];
      }
      s = dot;
    }
    const float pmax = wave_reduce_max(s);
    const float mn = fmaxf(fmaxf(m, pmax), -1e30f);
    const float alpha = exp2f((m - mn) * 1.4426950408889634f);
    const float p = (key < L)
        ? exp2f((s - mn) * 1.4426950408889634f) : 0.f;
    l = l * alpha + wave_reduce_sum(p);
    m = mn;
    p_lds[wid * 64 + lane] = p;
#pragma unroll
    for (int i = 0; i < DPL; ++i) acc[i] *= alpha;
    // branch-free 8-deep PV: independent loads pipeline over L2 latency
    const int kmax = min(64, L - t0);
    int kk = 0;
    for (; kk + 8 <= kmax; kk += 8) {
      float pv[8];
      float vv[8][DPL];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int key2 = t0 + kk + j;
        pv[j] = p_lds[wid * 64 + kk + j];
        const CT* vrow = vcache +
            (long)bt_lds[key2 / PBLK] * blk_stride + head_off +
            (key2 % PBLK) * D + lane * DPL;
#pragma unroll
        for (int i = 0; i < DPL; ++i)
          vv[j][i] = (sizeof(CT) == 2)
              ? bf2f((unsigned short)vrow[i])
              : fp8_to_f32((unsigned char)vrow[i]);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j)
#pragma unroll
        for (int i = 0; i < DPL; ++i) acc[i] += pv[j] * vv[j][i];
    }
    for (; kk < kmax; ++kk) {
      const int key2 = t0 + kk;
      const float pv = p_lds[wid * 64 + kk];
      const CT* vrow = vcache +
          (long)bt_lds[key2 / PBLK] * blk_stride + head_off +
          (key2 % PBLK) * D + lane * DPL;
#pragma unroll
      for (int i = 0; i < DPL; ++i)
        acc[i] += pv * ((sizeof(CT) == 2)
                        ? bf2f((unsigned short)vrow[i])
                        : fp8_to_f32((unsigned char)vrow[i]));
    }
  }

  if (lane == 0) {
    wstate[wid * 2] = m;
    wstate[wid * 2 + 1] = l;
  }
#pragma unroll
  for (int i = 0; i < DPL; ++i) wacc[wid * D + lane * DPL + i] = acc[i];
  __syncthreads();

  if (wid == 0) {
    float M = -1e30f;
#pragma unroll
    for (int w = 0; w < 4; ++w) M = fmaxf(M, wstate[w * 2]);
    float Ltot = 0.f;
    float o[2] = {0.f, 0.f};
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float a = exp2f((wstate[w * 2] - M) * 1.4426950408889634f);
      Ltot += wstate[w * 2 + 1] * a;
#pragma unroll
      for (int i = 0; i < DPL; ++i) o[i] += a * wacc[w * D + lane * DPL + i];
    }
    const float inv = (Ltot > 0.f) ? 1.f / Ltot : 0.f;
#pragma unroll
    for (int i = 0; i < DPL; ++i)
      out[((long)b * H + h) * D + lane * DPL + i] = f2bf(o[i] * inv);
  }
}

// GQA-grouped variant: ONE workgroup per (b, kv-head) computes all
// GR = H/HKV query heads that share that KV head, so each K/V row is
// read ONCE instead of GR times. At llama3-8b shapes (GR=4) this cuts
// decode-attention HBM traffic 4x; at bs128 ctx2048 the step is
// KV-bandwidth-bound, so it shows up ~1:1 end-to-end.
template <int D, int GR, typename CT>
__global__ void __launch_bounds__(256)
paged_attn_gqa_kernel(const unsigned short* __restrict__ q,
                      const CT* __restrict__ kcache,
                      const CT* __restrict__ vcache,
                      const int* __restrict__ block_table,
                      const int* __restrict__ ctx_len,
                      unsigned short* __restrict__ out, int B, int H,
                      int HKV, int max_blocks, float scale) {
  const int b = blockIdx.x / HKV;
  const int kvh = blockIdx.x % HKV;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int L = ctx_len[b];

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* p_lds = reinterpret_cast<float*>(smem);        // 4*GR*64
  float* q_lds = p_lds + 4 * GR * 64;                   // GR*D
  float* wstate = q_lds + GR * D;                       // 4*GR*2
  float* wacc = wstate + 8 * GR;                        // 4*GR*D
  int* bt_lds = reinterpret_cast<int*>(wacc + 4 * GR * D);

  const int nblk = (L + PBLK - 1) / PBLK;
  for (int i = threadIdx.x; i < nblk; i += 256)
    bt_lds[i] = block_table[(long)b * max_blocks + i];
  for (int i = threadIdx.x; i < GR * D; i += 256) {
    const int g = i / D, d = i % D;
    q_lds[i] = bf2f(q[((long)b * H + kvh * GR + g) * D + d]) * scale;
  }
  __syncthreads();

  constexpr int DPL = D / 64;
  float m[GR], l[GR], acc[GR][DPL];
#pragma unroll
  for (int g = 0; g < GR; ++g) {
    m[g] = -1e30f;
    l[g] = 0.f;
#pragma unroll
    for (int i = 0; i < DPL; ++i) acc[g][i] = 0.f;
  }
  const long blk_stride = (long)HKV * PBLK * D;
  const long head_off = (long)kvh * PBLK * D;

  for (int t0 = wid * 64; t0 < L; t0 += 256) {
    const int key = t0 + lane;
    float dot[GR];
#pragma unroll
    for (int g = 0; g < GR; ++g) dot[g] = 0.f;
    if (key < L) {
      const CT* krow = kcache +
          (long)bt_lds[key / PBLK] * blk_stride + head_off +
          (key % PBLK) * D;
#pragma unroll
      for (int c = 0; c < D / 8; ++c) {
        float kf[8];
        if constexpr (sizeof(CT) == 2) {
          s16x8 kv8 = *reinterpret_cast<const s16x8*>(krow + c * 8);
          bf8_to_f32(kv8, kf);
        } else {
          i32x2 kv8 = *reinterpret_cast<const i32x2*>(krow + c * 8);
          const unsigned char* bb =
              reinterpret_cast<const unsigned char*>(&kv8);
#pragma unroll
          for (int i = 0; i < 8; ++i) kf[i] = fp8_to_f32(bb[i]);
        }
#pragma unroll
        for (int g = 0; g < GR; ++g)
#pragma unroll
          for (int i = 0; i < 8; ++i)
            dot[g] += kf[i] * q_lds[g * D + c * 8 + i];
      }
    }
#pragma unroll
    for (int g = 0; g < GR; ++g) {
      const float s = (key < L) ? dot[g] : -INFINITY;
      const float pmax = wave_reduce_max(s);
      const float mn = fmaxf(fmaxf(m[g], pmax), -1e30f);
      const float alpha = exp2f((m[g] - mn) * 1.4426950408889634f);
      const float p = (key < L)
          ? exp2f((s - mn) * 1.4426950408889634f) : 0.f;
      l[g] = l[g] * alpha + wave_reduce_sum(p);
      m[g] = mn;
      p_lds[(wid * GR + g) * 64 + lane] = p;
#pragma unroll
      for (int i = 0; i < DPL; ++i) acc[g][i] *= alpha;
    }
    const int kmax = min(64, L - t0);
    int kk = 0;
    for (; kk + 4 <= kmax; kk += 4) {
      float vv[4][DPL];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int key2 = t0 + kk + j;
        const CT* vrow = vcache +
            (long)bt_lds[key2 / PBLK] * blk_stride + head_off +
            (key2 % PBLK) * D + lane * DPL;
#pragma unroll
        for (int i = 0; i < DPL; ++i)
          vv[j][i] = (sizeof(CT) == 2)
              ? bf2f((unsigned short)vrow[i])
              : fp8_to_f32((unsigned char)vrow[i]);
      }
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int g = 0; g < GR; ++g) {
          const float pv = p_lds[(wid * GR + g) * 64 + kk + j];
#pragma unroll
          for (int i = 0; i < DPL; ++i) acc[g][i] += pv * vv[j][i];
        }
    }
    for (; kk < kmax; ++kk) {
      const int key2 = t0 + kk;
      const CT* vrow = vcache +
          (long)bt_lds[key2 / PBLK] * blk_stride + head_off +
          (key2 % PBLK) * D + lane * DPL;
      float vv[DPL];
#pragma unroll
      for (int i = 0; i < DPL; ++i)
        vv[i] = (sizeof(CT) == 2)
            ? bf2f((unsigned short)vrow[i])
            : fp8_to_f32((unsigned char)vrow[i]);
#pragma unroll
      for (int g = 0; g < GR; ++g) {
        const float pv = p_lds[(wid * GR + g) * 64 + kk];
#pragma unroll
        for (int i = 0; i < DPL; ++i) acc[g][i] += pv * vv[i];
      }
    }
  }

#pragma unroll
  for (int g = 0; g < GR; ++g) {
    if (lane == 0) {
      wstate[(wid * GR + g) * 2] = m[g];
      wstate[(wid * GR + g) * 2 + 1] = l[g];
    }
#pragma unroll
    for (int i = 0; i < DPL; ++i)
      wacc[(wid * GR + g) * D + lane * DPL + i] = acc[g][i];
  }
  __syncthreads();

  if (wid == 0) {
#pragma unroll
    for (int g = 0; g < GR; ++g) {
      float M = -1e30f;
#pragma unroll
      for (int w = 0; w < 4; ++w)
        M = fmaxf(M, wstate[(w * GR + g) * 2]);
      float Ltot = 0.f;
      float o[DPL];
#pragma unroll
      for (int i = 0; i < DPL; ++i) o[i] = 0.f;
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        const float a =
            exp2f((wstate[(w * GR + g) * 2] - M) * 1.4426950408889634f);
        Ltot += wstate[(w * GR + g) * 2 + 1] * a;
#pragma unroll
        for (int i = 0; i < DPL; ++i)
          o[i] += a * wacc[(w * GR + g) * D + lane * DPL + i];
      }
      const float inv = (Ltot > 0.f) ? 1.f / Ltot : 0.f;
#pragma unroll
      for (int i = 0; i < DPL; ++i)
        out[((long)b * H + kvh * GR + g) * D + lane * DPL + i] =
            f2bf(o[i] * inv);
    }
  }
}

// Context-split ("flash-decoding") variant for SMALL batches: B*HKV
// workgroups underfill 256 CUs (B=1, HKV=8 -> 8 WGs = 3% of the chip),
// so split each sequence's context into NS chunks, compute partial
// (m, l, acc) per chunk in parallel, and combine in a second kernel.
// Scratch: pacc [B*HKV*NS, GR, D] f32 ; pml [B*HKV*NS, GR, 2] f32.
template <int D, int GR, typename CT>
__global__ void __launch_bounds__(256)
paged_attn_split_kernel(const unsigned short* __restrict__ q,
                        const CT* __restrict__ kcache,
                        const CT* __restrict__ vcache,
                        const int* __restrict__ block_table,
                        const int* __restrict__ ctx_len,
                        float* __restrict__ pacc, float* __restrict__ pml,
                        int B, int H, int HKV, int max_blocks, int NS,
                        float scale) {
  const int split = blockIdx.x % NS;
  const int bg = blockIdx.x / NS;         // (b * HKV + kvh)
  const int b = bg / HKV;
  const int kvh = bg % HKV;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int L = ctx_len[b];
  const int chunk = ((L + NS * PBLK - 1) / (NS * PBLK)) * PBLK;
  const int s0 = split * chunk;
  const int s1 = min(L, s0 + chunk);

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* p_lds = reinterpret_cast<float*>(smem);        // 4*GR*64
  float* q_lds = p_lds + 4 * GR * 64;                   // GR*D
  float* wstate = q_lds + GR * D;                       // 4*GR*2
  float* wacc = wstate + 8 * GR;                        // 4*GR*D
  int* bt_lds = reinterpret_cast<int*>(wacc + 4 * GR * D);

  const int nblk = (L + PBLK - 1) / PBLK;
  for (int i = threadIdx.x; i < nblk; i += 256)
    bt_lds[i] = block_table[(long)b * max_blocks + i];
  for (int i = threadIdx.x; i < GR * D; i += 256) {
    const int g = i / D, d = i % D;
    q_lds[i] = bf2f(q[((long)b * H + kvh * GR + g) * D + d]) * scale;
  }
  __syncthreads();

  constexpr int DPL = D / 64;
  float m[GR], l[GR], acc[GR][DPL];
#pragma unroll
  for (int g = 0; g < GR; ++g) {
    m[g] = -1e30f;
    l[g] = 0.f;
#pragma unroll
    for (int i = 0; i < DPL; ++i) acc[g][i] = 0.f;
  }
  const long blk_stride = (long)HKV * PBLK * D;
  const long head_off = (long)kvh * PBLK * D;

  for (int t0 = s0 + wid * 64; t0 < s1; t0 += 256) {
    const int key = t0 + lane;
    float dot[GR];
#pragma unroll
    for (int g = 0; g < GR; ++g) dot[g] = 0.f;
    if (key < s1) {
      const CT* krow = kcache +
          (long)bt_lds[key / PBLK] * blk_stride + head_off +
          (key % PBLK) * D;
#pragma unroll
      for (int c = 0; c < D / 8; ++c) {
        float kf[8];
        if constexpr (sizeof(CT) == 2) {
          s16x8 kv8 = *reinterpret_cast<const s16x8*>(krow + c * 8);
          bf8_to_f32(kv8, kf);
        } else {
          i32x2 kv8 = *reinterpret_cast<const i32x2*>(krow + c * 8);
          const unsigned char* bb =
              reinterpret_cast<const unsigned char*>(&kv8);
#pragma unroll
          for (int i = 0; i < 8; ++i) kf[i] = fp8_to_f32(bb[i]);
        }
#pragma unroll
        for (int g = 0; g < GR; ++g)
#pragma unroll
          for (int i = 0; i < 8; ++i)
            dot[g] += kf[i] * q_lds[g * D + c * 8 + i];
      }
    }
#pragma unroll
    for (int g = 0; g < GR; ++g) {
      const float s = (key < s1) ? dot[g] : -INFINITY;
      const float pmax = wave_reduce_max(s);
      const float mn = fmaxf(fmaxf(m[g], pmax), -1e30f);
      const float alpha = exp2f((m[g] - mn) * 1.4426950408889634f);
      const float p = (key < s1)
          ? exp2f((s - mn) * 1.4426950408889634f) : 0.f;
      l[g] = l[g] * alpha + wave_reduce_sum(p);
      m[g] = mn;
      p_lds[(wid * GR + g) * 64 + lane] = p;
#pragma unroll
      for (int i = 0; i < DPL; ++i) acc[g][i] *= alpha;
    }
    const int kmax = min(64, s1 - t0);
    for (int kk = 0; kk < kmax; ++kk) {
      const int key2 = t0 + kk;
      const CT* vrow = vcache +
          (long)bt_lds[key2 / PBLK] * blk_stride + head_off +
          (key2 % PBLK) * D + lane * DPL;
      float vv[DPL];
#pragma unroll
      for (int i = 0; i < DPL; ++i)
        vv[i] = (sizeof(CT) == 2)
            ? bf2f((unsigned short)vrow[i])
            : fp8_to_f32((unsigned char)vrow[i]);
#pragma unroll
      for (int g = 0; g < GR; ++g) {
        const float pv = p_lds[(wid * GR + g) * 64 + kk];
#pragma unroll
        for (int i = 0; i < DPL; ++i) acc[g][i] += pv * vv[i];
      }
    }
  }

#pragma unroll
  for (int g = 0; g < GR; ++g) {
    if (lane == 0) {
      wstate[(wid * GR + g) * 2] = m[g];
      wstate[(wid * GR + g) * 2 + 1] = l[g];
    }
#pragma unroll
    for (int i = 0; i < DPL; ++i)
      wacc[(wid * GR + g) * D + lane * DPL + i] = acc[g][i];
  }
  __syncthreads();

  if (wid == 0) {
    const long po = ((long)bg * NS + split) * GR;
#pragma unroll
    for (int g = 0; g < GR; ++g) {
      float M = -1e30f;
#pragma unroll
      for (int w = 0; w < 4; ++w)
        M = fmaxf(M, wstate[(w * GR + g) * 2]);
      float Ltot = 0.f;
      float o[DPL];
#pragma unroll
      for (int i = 0; i < DPL; ++i) o[i] = 0.f;
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        const float a =
            exp2f((wstate[(w * GR + g) * 2] - M) * 1.4426950408889634f);
        Ltot += wstate[(w * GR + g) * 2 + 1] * a;
#pragma unroll
        for (int i = 0; i < DPL; ++i)
          o[i] += a * wacc[(w * GR + g) * D + lane * DPL + i];
      }
#pragma unroll
      for (int i = 0; i < DPL; ++i)
        pacc[(po + g) * D + lane * DPL + i] = o[i];
      if (lane == 0) {
        pml[(po + g) * 2] = M;
        pml[(po + g) * 2 + 1] = Ltot;
      }
    }
  }
}

// Combine NS partials -> normalized bf16 out. Grid: B*HKV, 256 thr.
template <int D, int GR>
__global__ void __launch_bounds__(256)
paged_attn_combine_kernel(const float* __restrict__ pacc,
                          const float* __restrict__ pml,
                          unsigned short* __restrict__ out, int B, int H,
                          int HKV, int NS) {
  const int bg = blockIdx.x;
  const int b = bg / HKV;
  const int kvh = bg % HKV;
  constexpr int DPL = D / 64;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  // wave w handles head group g = wid (GR <= 4 per wave pass)
  for (int g = wid; g < GR; g += 4) {
    const long po = ((long)bg * NS) * GR;
    float M = -1e30f;
    for (int s = 0; s < NS; ++s)
      M = fmaxf(M, pml[(po + (long)s * GR + g) * 2]);
    float Ltot = 0.f;
    float o[DPL];
#pragma unroll
    for (int i = 0; i < DPL; ++i) o[i] = 0.f;
    for (int s = 0; s < NS; ++s) {
      const long pi = po + (long)s * GR + g;
      const float a = exp2f((pml[pi * 2] - M) * 1.4426950408889634f);
      Ltot += pml[pi * 2 + 1] * a;
#pragma unroll
      for (int i = 0; i < DPL; ++i)
        o[i] += a * pacc[pi * D + lane * DPL + i];
    }
    const float inv = (Ltot > 0.f) ? 1.f / Ltot : 0.f;
#pragma unroll
    for (int i = 0; i < DPL; ++i)
      out[((long)b * H + kvh * GR + g) * D + lane * DPL + i] =
          f2bf(o[i] * inv);
  }
}

extern "C" void launch_paged_attn_split(
    const void* q, const void* kcache, const void* vcache,
    const int* block_table, const int* ctx_len, void* pacc, void* pml,
    void* out, int B, int H, int HKV, int D, int max_blocks, int NS,
    float scale, bool fp8, hipStream_t stream) {
  const int GR = H / HKV;
#define PS(DD, GG, CT)                                                         \
  do {                                                                         \
    const int lds_g =                                                          \
        (4 * GG * 64 + GG * DD + 8 * GG + 4 * GG * DD) * 4 + max_blocks * 4;   \
    hipLaunchKernelGGL((paged_attn_split_kernel<DD, GG, CT>),                  \
                       dim3(B * HKV * NS), dim3(256), lds_g, stream,           \
                       (const unsigned short*)q, (const CT*)kcache,            \
                       (const CT*)vcache, block_table, ctx_len,                \
                       (float*)pacc, (float*)pml, B, H, HKV, max_blocks, NS,   \
                       scale);                                                 \
    hipLaunchKernelGGL((paged_attn_combine_kernel<DD, GG>),                    \
                       dim3(B * HKV), dim3(256), 0, stream,                    \
                       (const float*)pacc, (const float*)pml,                  \
                       (unsigned short*)out, B, H, HKV, NS);                   \
  } while (0)
#define PS_D(DD, CT)                                                           \
  do {                                                                         \
    if (GR == 1) PS(DD, 1, CT);                                                \
    else if (GR == 2) PS(DD, 2, CT);                                           \
    else if (GR == 4) PS(DD, 4, CT);                                           \
    else PS(DD, 8, CT);                                                        \
  } while (0)
  if (D == 64) {
    if (fp8) PS_D(64, unsigned char); else PS_D(64, unsigned short);
  } else if (D == 128) {
    if (fp8) PS_D(128, unsigned char); else PS_D(128, unsigned short);
  }
#undef PS_D
#undef PS
}

extern "C" void launch_paged_attn(const void* q, const void* kcache,
                                  const void* vcache, const int* block_table,
                                  const int* ctx_len, void* out, int B, int H,
                                  int HKV, int D, int max_blocks, float scale,
                                  bool fp8, hipStream_t stream) {
  const int GR = (HKV > 0 && H % HKV == 0) ? H / HKV : 1;
  if (GR == 2 || GR == 4 || GR == 8) {
    // grouped launch: each workgroup serves one (b, kv-head), K/V read
    // once for all GR query heads
    dim3 grid(B * HKV);
#define PG(DD, GG, CT)                                                         \
  do {                                                                         \
    const int lds_g =                                                          \
        (4 * GG * 64 + GG * DD + 8 * GG + 4 * GG * DD) * 4 + max_blocks * 4;   \
    hipLaunchKernelGGL((paged_attn_gqa_kernel<DD, GG, CT>), grid, dim3(256),   \
                       lds_g, stream, (const unsigned short*)q,                \
                       (const CT*)kcache, (const CT*)vcache, block_table,      \
                       ctx_len, (unsigned short*)out, B, H, HKV, max_blocks,   \
                       scale);                                                 \
  } while (0)
#define PG_D(DD, CT)                                                           \
  do {                                                                         \
    if (GR == 2) PG(DD, 2, CT);                                                \
    else if (GR == 4) PG(DD, 4, CT);                                           \
    else PG(DD, 8, CT);                                                        \
  } while (0)
    if (D == 64) {
      if (fp8) PG_D(64, unsigned char); else PG_D(64, unsigned short);
    } else if (D == 128) {
      if (fp8) PG_D(128, unsigned char); else PG_D(128, unsigned short);
    }
#undef PG_D
#undef PG
    return;
  }
  const int lds = (4 * 64 + D + 8 + 4 * D) * 4 + max_blocks * 4;
  dim3 grid(B * H);
#define PA(DD, CT)                                                             \
  hipLaunchKernelGGL((paged_attn_kernel<DD, CT>), grid, dim3(256), lds,        \
                     stream, (const unsigned short*)q, (const CT*)kcache,      \
                     (const CT*)vcache, block_table, ctx_len,                  \
                     (unsigned short*)out, B, H, HKV, max_blocks, scale)
  if (D == 64) {
    if (fp8) PA(64, unsigned char); else PA(64, unsigned short);
  } else if (D == 128) {
    if (fp8) PA(128, unsigned char); else PA(128, unsigned short);
  }
#undef PA
}

// Write one new token's K/V into the paged cache (decode step).
// knew/vnew: [B, HKV, D] bf16; slot[b] = ctx position of the new token.
template <int D, typename CT>
__global__ void __launch_bounds__(256)
cache_put_kernel(const unsigned short* __restrict__ knew,
                 const unsigned short* __restrict__ vnew,
                 CT* __restrict__ kcache,
                 CT* __restrict__ vcache,
                 const int* __restrict__ block_table,
                 const int* __restrict__ slot, int B, int HKV,
                 int max_blocks) {
  const long total = (long)B * HKV * D;
  const long stride = (long)gridDim.x * blockDim.x;
  const long blk_stride = (long)HKV * PBLK * D;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int d = (int)(i % D);
    const int kvh = (int)((i / D) % HKV);
    const int b = (int)(i / ((long)D * HKV));
    const int pos = slot[b];
    const int blk = block_table[(long)b * max_blocks + pos / PBLK];
    const long off =
        (long)blk * blk_stride + ((long)kvh * PBLK + pos % PBLK) * D + d;
    if constexpr (sizeof(CT) == 2) {
      kcache[off] = knew[i];
      vcache[off] = vnew[i];
    } else {
      kcache[off] = __hip_fp8_e4m3(bf2f(knew[i])).__x;
      vcache[off] = __hip_fp8_e4m3(bf2f(vnew[i])).__x;
    }
  }
}

extern "C" void launch_cache_put(const void* knew, const void* vnew,
                                 void* kcache, void* vcache,
                                 const int* block_table, const int* slot,
                                 int B, int HKV, int D, int max_blocks,
                                 bool fp8, hipStream_t stream) {
  const long total = (long)B * HKV * D;
  int blocks = (int)min((total + 255) / 256, (long)2048);
#define CP(DD, CT)                                                             \
  hipLaunchKernelGGL((cache_put_kernel<DD, CT>), dim3(blocks), dim3(256), 0,   \
                     stream, (const unsigned short*)knew,                      \
                     (const unsigned short*)vnew, (CT*)kcache, (CT*)vcache,    \
                     block_table, slot, B, HKV, max_blocks)
  if (D == 64) {
    if (fp8) CP(64, unsigned char); else CP(64, unsigned short);
  } else {
    if (fp8) CP(128, unsigned char); else CP(128, unsigned short);
  }
#undef CP
}
