// Hand-written CDNA4 MFMA GEMM: C[M,N] = A[M,K] @ B[N,K]^T (+bias, +act).
//
// This is the fused-epilogue GEMM for encoder FFN/projection layers
// (reference delegated all GEMMs to HF transformers / vLLM; SURVEY.md
// K1/K2). Plain unfused GEMMs may still go through hipBLASLt; this
// kernel exists for the fused bias+GELU/SiLU paths and as the
// MFMA/LDS-tiled evidence path.
//
// Structure (guide §5 anatomy + "minimum 2-phase" recipe):
//  - 128x128x64 tile, 256 threads = 4 waves, each wave a 64x64 subtile
//    as 4x4 fragments of mfma_f32_16x16x32_bf16 (fp32 accum).
//  - global->LDS staging via __builtin_amdgcn_global_load_lds width=16
//    (lane-linear dest), double-buffered.
//  - T2 XOR swizzle moved to the SOURCE address (rule 21): LDS image is
//    lane-linear; both the staging source chunk and the ds_read_b128
//    chunk are XORed with (row&7) -> <=2-way bank conflicts.
//  - T1 bijective XCD-aware blockIdx swizzle.
//  - Epilogue: accs staged through padded LDS (f32 [64][65] per wave),
//    fused bias + activation, coalesced bf16 row stores.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define BM 128
#define BN 128
#define BK 64
// LDS: A/B tiles double buffered: 2*2*128*64*2B = 64 KiB; epilogue
// reuses the same region (4 waves * 64*65*4B = 66.6 KiB) -> 68 KiB.
#define LDS_BYTES (68 * 1024)
#define TILE_BYTES (BM * BK * 2)  // 16 KiB

enum Act { ACT_NONE = 0, ACT_GELU_TANH = 1, ACT_GELU_ERF = 2, ACT_SILU = 3 };

extern __shared__ __attribute__((aligned(16))) char smem[];

// Stage one 128x64 bf16 tile (A or B quarter per wave) with glds width 16.
// src: global base of the tile (row 0, k 0), row stride = K elements.
// lds_off: byte offset of the tile buffer inside smem.
__device__ __forceinline__ void stage_tile(const unsigned short* __restrict__ g,
                                           long row_stride, int lds_off,
                                           int wid, int lane) {
  // wave w covers rows [32w, 32w+32) as 4 pieces of 8 rows; each piece:
  // lane l -> row = 8p + (l>>3), dst chunk = l&7, src chunk = (l&7)^(l>>3).
  const int r_in_piece = lane >> 3;
  const int c_dst = lane & 7;
  const int c_src = c_dst ^ r_in_piece;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int row = wid * 32 + p * 8 + r_in_piece;
    const unsigned short* src = g + (long)row * row_stride + c_src * 8;
    auto dst = (__attribute__((address_space(3))) unsigned int*)(
        smem + lds_off + (wid * 32 + p * 8) * 128 + lane * 16);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src, dst, 16, 0,
        0);
  }
}

// Read one MFMA A/B fragment from a swizzled tile image.
// row: tile-local row (0..127); chunk: k/8 (0..7).
__device__ __forceinline__ bf16x8 read_frag(int lds_off, int row, int chunk) {
  const int swz = chunk ^ (row & 7);
  return *reinterpret_cast<const bf16x8*>(smem + lds_off + row * 128 +
                                          swz * 16);
}

template <int ACT, bool HAS_BIAS>
__global__ void __launch_bounds__(256, 1)
gemm_bf16_kernel(const unsigned short* __restrict__ A,
                 const unsigned short* __restrict__ B,
                 const unsigned short* __restrict__ bias,
                 unsigned short* __restrict__ C, int M, int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;

  // T1: bijective XCD swizzle of the linear workgroup id.
  const int ntm = M / BM, ntn = N / BN;
  const int nwg = ntm * ntn;
  int bid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, j = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + j;
  }
  const int tm = bid / ntn, tn = bid % ntn;

  const unsigned short* Atile = A + (long)tm * BM * K;
  const unsigned short* Btile = B + (long)tn * BN * K;

  const int bufA[2] = {0, TILE_BYTES};
  const int bufB[2] = {2 * TILE_BYTES, 3 * TILE_BYTES};

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int wrow = (wid >> 1) * 64;  // wave's 64x64 subtile origin
  const int wcol = (wid & 1) * 64;

  const int nt = K / BK;
  int cur = 0;
  stage_tile(Atile, K, bufA[0], wid, lane);
  stage_tile(Btile, K, bufB[0], wid, lane);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < nt; ++t) {
    if (t + 1 < nt) {
      stage_tile(Atile + (long)(t + 1) * BK, K, bufA[cur ^ 1], wid, lane);
      stage_tile(Btile + (long)(t + 1) * BK, K, bufB[cur ^ 1], wid, lane);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[4], bf[4];
      const int chunk = kk * 4 + (lane >> 4);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        af[mi] = read_frag(bufA[cur], wrow + mi * 16 + (lane & 15), chunk);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bf[ni] = read_frag(bufB[cur], wcol + ni * 16 + (lane & 15), chunk);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: stage accs through padded LDS, fuse bias+act ----------
  float* ep = reinterpret_cast<float*>(smem) + wid * 64 * 65;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      // C/D map: col = lane&15, row = (lane>>4)*4 + reg
      const int col = ni * 16 + (lane & 15);
      const int row0 = mi * 16 + (lane >> 4) * 4;
#pragma unroll
      for (int r = 0; r < 4; ++r) ep[(row0 + r) * 65 + col] = acc[mi][ni][r];
    }
  __syncthreads();

  const int grow0 = tm * BM + wrow;
  const int gcol0 = tn * BN + wcol;
  float bv = 0.f;
  if (HAS_BIAS) bv = bf2f(bias[gcol0 + lane]);
  for (int r = 0; r < 64; ++r) {
    float v = ep[r * 65 + lane];
    if (HAS_BIAS) v += bv;
    if (ACT == ACT_GELU_TANH) v = fast_gelu(v);
    else if (ACT == ACT_GELU_ERF) v = gelu_erf(v);
    else if (ACT == ACT_SILU) v = silu(v);
    C[(long)(grow0 + r) * N + gcol0 + lane] = f2bf(v);
  }
}

extern "C" void launch_gemm_bf16(const void* A, const void* B,
                                 const void* bias, void* C, int M, int N,
                                 int K, int act, hipStream_t stream) {
  const int nwg = (M / BM) * (N / BN);
#define GD(ACT, HB)                                                            \
  hipLaunchKernelGGL((gemm_bf16_kernel<ACT, HB>), dim3(nwg), dim3(256),        \
                     LDS_BYTES, stream, (const unsigned short*)A,              \
                     (const unsigned short*)B, (const unsigned short*)bias,    \
                     (unsigned short*)C, M, N, K)
  const bool hb = bias != nullptr;
  switch (act) {
    case ACT_GELU_TANH:
      if (hb) GD(1, true); else GD(1, false);
      break;
    case ACT_GELU_ERF:
      if (hb) GD(2, true); else GD(2, false);
      break;
    case ACT_SILU:
      if (hb) GD(3, true); else GD(3, false);
      break;
    default:
      if (hb) GD(0, true); else GD(0, false);
      break;
  }
#undef GD
}
