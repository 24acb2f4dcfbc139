// Deep-pipelined CDNA4 MFMA GEMM: C[M,N] = A[M,K] @ B[N,K]^T (+bias/act).
// Replaces the dense projection GEMMs inside the reference's HF/vLLM
// model forwards (reference distllm/embed/encoders/auto.py encode();
// SURVEY.md §2.9 K1/K2 rows) for shapes where it beats hipBLASLt —
// the per-shape winner is picked by runtime autotune in ops.linear.
//
// 256x256x64 tile, 512 threads = 8 waves, each wave computes ALL 256
// rows x 32 cols (16x2 fragments of mfma_f32_16x16x32_bf16). This
// wave->column mapping is what makes a >1-tile-deep prefetch provable
// with double buffering: the A row-half an entire phase-pair stops
// reading frees mid-tile, so A halves stage TWO tiles ahead and the
// inter-tile wait is a counted vmcnt, never a drain.
//
// Schedule per K-tile t (cur = buf[t&1], oth = cur^1; guide §5
// phased-template family, own derivation):
//   ph0: ds_read A-rh0(16) + B-all(4); stage oth.A1<-A1(t+1),
//        oth.B0<-B0(t+1); 16 MFMA (rh0, ch0);  [mid barrier]
//   ph1: stage oth.B1<-B1(t+1), cur.A0<-A0(t+2); 16 MFMA (rh0, ch1);
//   ph2: ds_read A-rh1(16); 16 MFMA (rh1, ch0);
//   ph3: 16 MFMA (rh1, ch1); s_waitcnt vmcnt(2) (leaves only A0(t+2)
//        in flight);  [boundary barrier]
// XOR swizzle chunk^(row&7) on the glds SOURCE address (rule 21);
// s_setprio(1) around MFMA clusters; raw s_barrier (one compiler
// memory fence per barrier, no vmcnt drains); fused bias+activation
// epilogue through padded-LDS transpose rounds.
// Requires M%256==0, N%256==0, K%64==0 (host falls back otherwise).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define G8_BM 256
#define G8_BN 256
#define G8_BK 64
#define HTILE_B (128 * G8_BK * 2)  // one half image: 16 KiB
#define G8_LDS (8 * HTILE_B)       // A/B x 2 buf x 2 half = 128 KiB

extern __shared__ __attribute__((aligned(16))) char smem[];

enum G8Act { G8_NONE = 0, G8_GELU_TANH = 1, G8_GELU_ERF = 2, G8_SILU = 3 };

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
  const int wid = threadIdx.x >> 6;   // 0..7: wave's 32-col slice

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

  auto img = [&](int op, int buf, int half) {
    return smem + ((op * 2 + buf) * 2 + half) * HTILE_B;
  };

  // stage one 128x64 half (16 glds; 2 per wave), swizzled source
  const int r_in = lane >> 3;
  const int c_dst = lane & 7;
  auto stage_half = [&](const unsigned short* G, int k0, int op, int buf,
                        int half) {
#pragma unroll
    for (int pp = 0; pp < 2; ++pp) {
      const int piece = wid * 2 + pp;
      const int row = piece * 8 + r_in;
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

  // fragment read: (row16 = frag row-block within the half, kk)
  const int fr_row = lane & 15;
  const int fr_c = lane >> 4;
  auto read_frag = [&](const char* image, int row16, int kk) {
    return *reinterpret_cast<const bf16x8*>(
        image + g8_img_off(row16 * 16 + fr_row, kk * 4 + fr_c));
  };

  f32x4 acc[16][2];
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    acc[i][0] = {0.f, 0.f, 0.f, 0.f};
    acc[i][1] = {0.f, 0.f, 0.f, 0.f};
  }

  const int nt = K / G8_BK;
  // prologue: all of tile 0 + A-half0 of tile 1
  stage_half(Atile, 0, 0, 0, 0);
  stage_half(Atile, 0, 0, 0, 1);
  stage_half(Btile, 0, 1, 0, 0);
  stage_half(Btile, 0, 1, 0, 1);
  if (nt > 1) stage_half(Atile, G8_BK, 0, 1, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  bf16x8 afr[16];  // 8 mi (current row-half) x 2 kk
  bf16x8 bfr[4];   // 2 ni x 2 kk (whole tile)
  // wave's B rows within its half image: (wid&3)*32 + ni*16 + fr_row
  const int bh = wid >> 2;        // B half index
  const int brow0 = (wid & 3) * 32;

  for (int t = 0; t < nt; ++t) {
    const int buf = t & 1;
    const char* Ai0 = (const char*)img(0, buf, 0);
    const char* Ai1 = (const char*)img(0, buf, 1);
    const char* Bi = (const char*)img(1, buf, bh);
    const int k1 = (t + 1) * G8_BK;
    const int k2 = (t + 2) * G8_BK;
    const bool n1 = t + 1 < nt, n2 = t + 2 < nt;

    // ---- ph0: reads (A rh0 + B all), stage oth.A1 + oth.B0, MFMA ----
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        bfr[ni * 2 + kk] =
            read_frag(Bi, (brow0 >> 4) + ni, kk);
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        afr[mi * 2 + kk] = read_frag(Ai0, mi, kk);
    if (n1) {
      stage_half(Atile, k1, 0, buf ^ 1, 1);
      stage_half(Btile, k1, 1, buf ^ 1, 0);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        acc[mi][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[mi * 2 + kk], bfr[kk], acc[mi][0], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("" ::: "memory");
    __builtin_amdgcn_s_barrier();   // mid barrier: A-rh0 reads done
    asm volatile("" ::: "memory");

    // ---- ph1: stage oth.B1 + cur.A0(t+2), MFMA (rh0, ch1) -----------
    if (n1) stage_half(Btile, k1, 1, buf ^ 1, 1);
    if (n2) stage_half(Atile, k2, 0, buf, 0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        acc[mi][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[mi * 2 + kk], bfr[2 + kk], acc[mi][1], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph2: reads A rh1, MFMA (rh1, ch0) --------------------------
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        afr[mi * 2 + kk] = read_frag(Ai1, mi, kk);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        acc[8 + mi][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[mi * 2 + kk], bfr[kk], acc[8 + mi][0], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph3: MFMA (rh1, ch1); counted wait; boundary barrier -------
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        acc[8 + mi][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[mi * 2 + kk], bfr[2 + kk], acc[8 + mi][1], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    if (n2) {
      // leave only A0(t+2) (the 2 newest glds) in flight
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    asm volatile("" ::: "memory");
    __builtin_amdgcn_s_barrier();   // tile boundary
    asm volatile("" ::: "memory");
  }

  // ---- epilogue: 4 rounds of 64 rows x 32 cols through padded LDS ----
  float* slab = reinterpret_cast<float*>(smem) + wid * 64 * 33;
  const int grow0 = tm * G8_BM;
  const int gcol0 = tn * G8_BN + wid * 32;
  float bv = 0.f;
  if (HAS_BIAS) bv = bf2f(bias[gcol0 + (lane & 31)]);
#pragma unroll
  for (int qd = 0; qd < 4; ++qd) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const f32x4 v = acc[qd * 4 + mi][ni];
        const int col = ni * 16 + (lane & 15);
        const int row0 = mi * 16 + (lane >> 4) * 4;
#pragma unroll
        for (int r = 0; r < 4; ++r) slab[(row0 + r) * 33 + col] = v[r];
      }
    __syncthreads();
    for (int rr = 0; rr < 32; ++rr) {
      const int r = rr * 2 + (lane >> 5);
      float v = slab[r * 33 + (lane & 31)];
      if (HAS_BIAS) v += bv;
      if (ACT == G8_GELU_TANH) v = fast_gelu(v);
      else if (ACT == G8_GELU_ERF) v = gelu_erf(v);
      else if (ACT == G8_SILU) v = silu(v);
      C[(long)(grow0 + qd * 64 + r) * N + gcol0 + (lane & 31)] = f2bf(v);
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
