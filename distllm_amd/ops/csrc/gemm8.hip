// Deep-pipelined CDNA4 MFMA GEMM: C[M,N] = A[M,K] @ B[N,K]^T (+bias/act).
//
// 256x256x64 tile, 512 threads = 8 waves (2M x 4N), each wave a 128x64
// C tile as 8x4 fragments of mfma_f32_16x16x32_bf16. Structure follows
// the guide's 256^2 phased template family (§5 "8-phase"), adapted to a
// schedule whose staging-overwrite and wait placement are provable:
//  - double-buffered A/B LDS images ([2 halves][128][64] bf16 each,
//    XOR-swizzled chunk ^ (row&7) against ds_read_b128 conflicts;
//    swizzle applied on the glds SOURCE address, rule 21);
//  - 4 phases per K-tile: each phase computes one C-quadrant (64x32,
//    16 MFMA) and stages ONE half-tile of the NEXT K-tile into the
//    other buffer (2 glds/wave);
//  - raw s_barrier between phases (no vmcnt(0) drain: the only VM wait
//    is ONE counted vmcnt(2) per K-tile, placed after the first phase's
//    stage issue — it covers all of the tile's own halves and leaves
//    the newest half-tile in flight);
//  - s_setprio(1) around each MFMA cluster (T5);
//  - epilogue: per-quadrant LDS transpose rounds -> fused bias +
//    activation -> coalesced bf16 stores.
// Requires M%256==0, N%256==0, K%64==0 (host pads / falls back).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define G8_BM 256
#define G8_BN 256
#define G8_BK 64
#define HTILE_B (128 * G8_BK * 2)  // one half image: 16 KiB
// LDS: A[2buf][2half] + B[2buf][2half] = 8 * 16 KiB = 128 KiB
#define G8_LDS (8 * HTILE_B)

extern __shared__ __attribute__((aligned(16))) char smem[];

enum G8Act { G8_NONE = 0, G8_GELU_TANH = 1, G8_GELU_ERF = 2, G8_SILU = 3 };

// image byte offset for element (row 0..127, chunk16 0..7) with swizzle
__device__ __forceinline__ int g8_img_off(int row, int c16) {
  return row * 128 + ((c16 ^ (row & 7)) * 16);
}

template <int ACT, bool HAS_BIAS>
__global__ void __launch_bounds__(512, 1)
gemm8_kernel(const unsigned short* __restrict__ A,
             const unsigned short* __restrict__ B,
             const unsigned short* __restrict__ bias,
             unsigned short* __restrict__ C, int M, int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;   // 0..7
  const int wm = wid >> 2;            // 0..1 (M half)
  const int wn = wid & 3;             // 0..3 (N quarter)

  // T1 bijective XCD swizzle
  const int ntm = M / G8_BM, ntn = N / G8_BN;
  const int nwg = ntm * ntn;
  int bid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, j = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + j;
  }
  const int tm = bid / ntn, tn = bid % ntn;

  const unsigned short* Atile = A + (long)tm * G8_BM * K;
  const unsigned short* Btile = B + (long)tn * G8_BN * K;

  // LDS half images: [op][buf][half]
  auto img = [&](int op, int buf, int half) {
    return smem + ((op * 2 + buf) * 2 + half) * HTILE_B;
  };

  // ---- staging: one half-tile (128 rows x 64 cols) by all 8 waves ----
  // piece = 8 rows (1 KiB glds); wave w takes pieces 2w, 2w+1.
  const int r_in = lane >> 3;          // row within piece
  const int c_dst = lane & 7;          // dest chunk16
  auto stage_half = [&](const unsigned short* G, int k0, int op, int buf,
                        int half) {
#pragma unroll
    for (int pp = 0; pp < 2; ++pp) {
      const int piece = wid * 2 + pp;
      const int row = piece * 8 + r_in;          // 0..127 within half
      const int c_src = c_dst ^ (row & 7);
      const unsigned short* src =
          G + (long)(half * 128 + row) * K + k0 + c_src * 8;
      auto dst = (__attribute__((address_space(3))) unsigned int*)(
          img(op, buf, half) + piece * 1024 + lane * 16);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src, dst,
          16, 0, 0);
    }
  };
  auto stage_tile_half = [&](int k0, int buf, int h) {
    // h: 0=A0, 1=B0, 2=A1, 3=B1 (A half first: consumed first)
    if (h == 0) stage_half(Atile, k0, 0, buf, 0);
    else if (h == 1) stage_half(Btile, k0, 1, buf, 0);
    else if (h == 2) stage_half(Atile, k0, 0, buf, 1);
    else stage_half(Btile, k0, 1, buf, 1);
  };

  // ---- fragment reads ------------------------------------------------
  // A-frag (mi 0..7 within wave's 128 rows, kk 0..1):
  //   row = mi*16 + (lane&15), chunk16 = kk*4 + (lane>>4)
  const int fr_row = lane & 15;
  const int fr_c = lane >> 4;  // 0..3
  auto read_frag = [&](const char* image, int row16, int kk) {
    return *reinterpret_cast<const bf16x8*>(
        image + g8_img_off(row16 * 16 + fr_row, kk * 4 + fr_c));
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nt = K / G8_BK;
  // prologue: tile 0 fully staged, synchronous
  for (int h = 0; h < 4; ++h) stage_tile_half(0, 0, h);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  bf16x8 bfr[8];  // B frags: 4 nfrag x 2 kk (all cached per tile)
  bf16x8 afr[8];  // A frags: 4 mi (current row-half) x 2 kk

  for (int t = 0; t < nt; ++t) {
    const int buf = t & 1;
    const char* Ai = (const char*)img(0, buf, wm);
    const char* Bi = (const char*)img(1, buf, wn >> 1);
    const int bn0 = (wn & 1) * 4;  // wave's 64 cols: nfrag 0..3 within
                                   // its B-half (128 rows of B image)
    const bool have_next = t + 1 < nt;
    const int k1 = (t + 1) * G8_BK;

#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      // stage the next tile into the other buffer: 2 halves in ph0 and
      // 2 in ph1, so by the vmcnt(0) at the END of ph3 every glds has
      // had >= 2 phases (~1100 cycles of MFMA) to land.
      if (have_next && ph == 0) {
        stage_tile_half(k1, buf ^ 1, 0);
        stage_tile_half(k1, buf ^ 1, 1);
      } else if (have_next && ph == 1) {
        stage_tile_half(k1, buf ^ 1, 2);
        stage_tile_half(k1, buf ^ 1, 3);
      }
      if (ph == 0) {
        // B frags for the whole tile + A frags of row-half 0
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            bfr[ni * 2 + kk] = read_frag(Bi, bn0 + ni, kk);
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            afr[mi * 2 + kk] = read_frag(Ai, mi, kk);
      } else if (ph == 2) {
        // A frags of row-half 1
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            afr[mi * 2 + kk] = read_frag(Ai, 4 + mi, kk);
      }
      // quadrant (rh = ph>>1, ch = ph&1): 4 mi x 2 ni x 2 kk MFMAs
      const int rh = ph >> 1;
      const int ch = ph & 1;
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[rh * 4 + mi][ch * 2 + ni] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afr[mi * 2 + kk], bfr[(ch * 2 + ni) * 2 + kk],
                    acc[rh * 4 + mi][ch * 2 + ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (ph == 3) {
        // every wave drains its own glds for tile t+1 BEFORE the
        // barrier, so the next tile's cross-wave reads are safe.
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      // compiler memory fences around the raw barrier: keep LDS
      // accesses on their side of it (s_barrier alone is not a fence)
      asm volatile("" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
    }
  }
  // drain remaining staged-but-unused glds (none: last tile stages none)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  // ---- epilogue: 4 quadrant rounds through LDS ------------------------
  // round q: each wave writes its quadrant rows q*32..q*32+31 (2 mfrag
  // x 4 nfrag) as f32 [32][65] into its 8.3KiB slab, then streams it
  // out coalesced with fused bias+act.
  float* slab = reinterpret_cast<float*>(smem) + wid * 32 * 65;
  const int grow0 = tm * G8_BM + wm * 128;
  const int gcol0 = tn * G8_BN + wn * 64;
  float bv = 0.f;
  if (HAS_BIAS) bv = bf2f(bias[gcol0 + (lane & 63)]);
#pragma unroll
  for (int qd = 0; qd < 4; ++qd) {
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const f32x4 v = acc[qd * 2 + mi][ni];
        const int col = ni * 16 + (lane & 15);
        const int row0 = mi * 16 + (lane >> 4) * 4;
#pragma unroll
        for (int r = 0; r < 4; ++r) slab[(row0 + r) * 65 + col] = v[r];
      }
    __syncthreads();
    // stream 32 rows x 64 cols: lane covers one col, rows sequential
    for (int r = 0; r < 32; ++r) {
      float v = slab[r * 65 + lane];
      if (HAS_BIAS) v += bv;
      if (ACT == G8_GELU_TANH) v = fast_gelu(v);
      else if (ACT == G8_GELU_ERF) v = gelu_erf(v);
      else if (ACT == G8_SILU) v = silu(v);
      C[(long)(grow0 + qd * 32 + r) * N + gcol0 + lane] = f2bf(v);
    }
    __syncthreads();
  }
}

extern "C" void launch_gemm8_bf16(const void* A, const void* B,
                                  const void* bias, void* C, int M, int N,
                                  int K, int act, hipStream_t stream) {
  const int nwg = (M / G8_BM) * (N / G8_BN);
#define G8D(ACT, HB)                                                           \
  hipLaunchKernelGGL((gemm8_kernel<ACT, HB>), dim3(nwg), dim3(512), G8_LDS,    \
                     stream, (const unsigned short*)A,                         \
                     (const unsigned short*)B, (const unsigned short*)bias,    \
                     (unsigned short*)C, M, N, K)
  const bool hb = bias != nullptr;
  switch (act) {
    case G8_GELU_TANH: if (hb) G8D(1, true); else G8D(1, false); break;
    case G8_GELU_ERF: if (hb) G8D(2, true); else G8D(2, false); break;
    case G8_SILU: if (hb) G8D(3, true); else G8D(3, false); break;
    default: if (hb) G8D(0, true); else G8D(0, false); break;
  }
#undef G8D
}
