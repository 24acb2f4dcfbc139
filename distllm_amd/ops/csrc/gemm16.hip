// 8-phase deep-pipelined CDNA4 MFMA GEMM: C[M,N] = A[M,K] @ B[N,K]^T
// (+fused bias/activation). The gemm8.hip successor targeting the
// K-shallow flagship encoder shapes (M=262144, N in {768,2304,3072},
// K in {768,3072}) where hipBLASLt previously owned ~52% of the
// flagship step (reference anchor: the delegated projection GEMMs of
// distllm/embed/encoders/auto.py:116-138).
//
// Structure per the guide's 256^2 8-phase template
// (cdna_hip_programming.md §5 "The 256² 8-phase template"), own
// implementation and schedule:
//   - 256x256x64 tile, 512 threads = 8 waves as 2(M) x 4(N); per-wave
//     output 128x64 (8 mi x 4 ni fragments of mfma_f32_16x16x32_bf16).
//   - one iteration consumes TWO K-tiles (8 phases); ZIGZAG quadrant
//     order {q(mi0-3,ni01), q(mi0-3,ni23), q(mi4-7,ni23),
//     q(mi4-7,ni01)}: at most one A-subtile (32 VGPR) plus both
//     B-subtiles (32) live at once -> no spills at 2 waves/SIMD.
//   - ONE barrier per phase: {ds_read subtile; glds stage;
//     s_barrier; lgkmcnt(0); s_setprio(1); 16 MFMA; s_setprio(0)}.
//     PMC on the two-barrier variant showed 37% of wave cycles parked
//     (SQ_WAIT_ANY); the post-MFMA barrier is dropped by DEFERRING
//     every stage slot to >= 2 barriers after its target image's last
//     read: any wave issuing the stage has passed BAR(p-1), and every
//     wave's reads from 2+ phases ago completed before its own
//     lgkmcnt(0)+MFMA which precede BAR(p-1) — placement-independent.
//   - stage slots (tile pair t,t+1 consuming bufs 0,1):
//       ph0: B1(t+1)+A0(t+1) -> buf1   ph1: A1(t+1) -> buf1
//       ph3: B0(t+2) -> buf0           ph4: B1(t+2)+A0(t+2) -> buf0
//       ph5: A1(t+2) -> buf0           ph7: B0(t+3) -> buf1
//     counted s_waitcnt vmcnt(2) ONCE per K-tile (end of ph3 and ph7)
//     leaves exactly the newest half-tile in flight (T3+T4).
//   - XOR swizzle (chunk ^ (row&7)) on the glds SOURCE address and the
//     ds_read offset (guide rule 21 / T2).
//   - T1 bijective XCD swizzle, tm-major so the N-tiles sharing an
//     A-panel land on one XCD's L2.
//   - epilogue: per-wave padded-LDS transpose rounds (wave-local
//     lgkmcnt ordering, no workgroup barriers); each lane packs 8 f32
//     into ONE 16-byte dwordx4 store per row pass (T21), bias +
//     branch-free GELU/SiLU fused.
// Requires M%256==0, N%256==0, K%128==0 (host falls back otherwise).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define G16_BM 256
#define G16_BN 256
#define G16_BK 64
#define G16_HTILE_B (128 * G16_BK * 2)  // one half image: 16 KiB
#define G16_LDS (8 * G16_HTILE_B)       // 2 buf x {A,B} x 2 half = 128 KiB

extern __shared__ __attribute__((aligned(16))) char smem16[];

enum G16Act { G16_NONE = 0, G16_GELU_TANH = 1, G16_GELU_ERF = 2,
              G16_SILU = 3 };

__device__ __forceinline__ int g16_img_off(int row, int c16) {
  return row * 128 + ((c16 ^ (row & 7)) * 16);
}

template <int ACT, bool HAS_BIAS>
__global__ void __launch_bounds__(512, 2)
gemm16_kernel(const unsigned short* __restrict__ A,
              const unsigned short* __restrict__ B,
              const unsigned short* __restrict__ bias,
              unsigned short* __restrict__ C, int M, int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;  // 0..7
  const int mw = wid >> 2;           // M half (0/1): rows mw*128..+127
  const int nw = wid & 3;            // N quarter: cols nw*64..+63

  // T1 bijective XCD swizzle (tm-major tile order underneath)
  const int ntm = M / G16_BM, ntn = N / G16_BN;
  const int nwg = ntm * ntn;
  int bid = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, j = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + j;
  }
  const int tm = bid / ntn, tn = bid % ntn;

  const unsigned short* Atile = A + (long)tm * G16_BM * K;
  const unsigned short* Btile = B + (long)tn * G16_BN * K;

  auto img = [&](int op, int buf, int half) {
    return smem16 + ((op * 2 + buf) * 2 + half) * G16_HTILE_B;
  };

  const int r_in = lane >> 3;   // 0..7 within the 8-row piece
  const int c_dst = lane & 7;   // 16B chunk within the row
  auto stage_half = [&](const unsigned short* G, long k0, int op, int buf,
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

  const int fr_row = lane & 15;
  const int fr_c = lane >> 4;  // kk sub-chunk 0..3 (8 bf16 each)
  auto read_frag = [&](const char* image, int row16, int kk) {
    return *reinterpret_cast<const bf16x8*>(
        image + g16_img_off(row16 * 16 + fr_row, kk * 4 + fr_c));
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int bh = nw >> 1;         // wave's B half image
  const int br16 = (nw & 1) * 4;  // local row16 base within it
  const int nt = K / G16_BK;      // even (host guarantees K%128==0)

  // ---- prologue: tile 0 complete + B0(1); steady state stages the
  // rest of tile 1 at ph0/ph1 of the first iteration ----------------
  stage_half(Atile, 0, 0, 0, 0);
  stage_half(Atile, 0, 0, 0, 1);
  stage_half(Btile, 0, 1, 0, 0);
  stage_half(Btile, 0, 1, 0, 1);
  stage_half(Btile, G16_BK, 1, 1, 0);
  asm volatile("s_waitcnt vmcnt(2)" ::: "memory");  // tile 0 landed
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  bf16x8 af[8];   // current A subtile: 4 mi x 2 kk
  bf16x8 b01[4];  // ni 0..1 x kk 0..1 (live the whole tile)
  bf16x8 b23[4];  // ni 2..3 x kk 0..1

  auto mfma_quad = [&](int mi0, int ni0, const bf16x8* a,
                       const bf16x8* b) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[mi0 + mi][ni0 + ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi * 2 + kk], b[ni * 2 + kk], acc[mi0 + mi][ni0 + ni],
              0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };
  auto rd_a = [&](const char* Ai, int mih) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        af[mi * 2 + kk] = read_frag(Ai, mih * 4 + mi, kk);
  };
  auto rd_b = [&](const char* Bi, int nih, bf16x8* dst) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        dst[ni * 2 + kk] = read_frag(Bi, br16 + nih * 2 + ni, kk);
  };

// one barrier per phase, placed between {reads+stage} and {MFMA}
#define G16_BAR()                                        \
  __builtin_amdgcn_s_barrier();                          \
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")

  const char* A0c = (const char*)img(0, 0, mw);  // wave's A half, buf0
  const char* B0c = (const char*)img(1, 0, bh);
  const char* A1c = (const char*)img(0, 1, mw);  // buf1
  const char* B1c = (const char*)img(1, 1, bh);

  const int niters = nt / 2;
  for (int it = 0; it < niters; ++it) {
    const int t = 2 * it;  // first K-tile of the pair (buf 0)
    const long kT1 = (long)(t + 1) * G16_BK;
    const long kT2 = (long)(t + 2) * G16_BK;
    const long kT3 = (long)(t + 3) * G16_BK;
    const bool hT2 = t + 2 < nt, hT3 = t + 3 < nt;

    // ======== K-tile t (buf 0); zigzag q00->q01->q11->q10 ========
    // ph0: read A(mi0-3)+B(ni0-1); stage B1(t+1)+A0(t+1) -> buf1
    rd_a(A0c, 0);
    rd_b(B0c, 0, b01);
    stage_half(Btile, kT1, 1, 1, 1);
    stage_half(Atile, kT1, 0, 1, 0);
    G16_BAR();
    mfma_quad(0, 0, af, b01);

    // ph1: read B(ni2-3); stage A1(t+1) -> buf1
    rd_b(B0c, 1, b23);
    stage_half(Atile, kT1, 0, 1, 1);
    G16_BAR();
    mfma_quad(0, 2, af, b23);

    // ph2: read A(mi4-7)
    rd_a(A0c, 1);
    G16_BAR();
    mfma_quad(4, 2, af, b23);

    // ph3: stage B0(t+2) -> buf0 (B(buf0) last read ph1: 2 bars ago);
    //      counted wait ensures tile t+1 fully landed for ph4
    if (hT2) stage_half(Btile, kT2, 1, 0, 0);
    __builtin_amdgcn_s_barrier();
    mfma_quad(4, 0, af, b01);
    if (hT2)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    asm volatile("" ::: "memory");
    __builtin_amdgcn_s_barrier();  // tile boundary (keeps ph4 reads
    asm volatile("" ::: "memory"); // behind every wave's ph3 MFMA srcs)

    // ======== K-tile t+1 (buf 1) ========
    // ph4: read A(mi0-3)+B(ni0-1); stage B1(t+2)+A0(t+2) -> buf0
    rd_a(A1c, 0);
    rd_b(B1c, 0, b01);
    if (hT2) {
      stage_half(Btile, kT2, 1, 0, 1);
      stage_half(Atile, kT2, 0, 0, 0);
    }
    G16_BAR();
    mfma_quad(0, 0, af, b01);

    // ph5: read B(ni2-3); stage A1(t+2) -> buf0
    rd_b(B1c, 1, b23);
    if (hT2) stage_half(Atile, kT2, 0, 0, 1);
    G16_BAR();
    mfma_quad(0, 2, af, b23);

    // ph6: read A(mi4-7)
    rd_a(A1c, 1);
    G16_BAR();
    mfma_quad(4, 2, af, b23);

    // ph7: stage B0(t+3) -> buf1; counted wait for tile t+2
    if (hT3) stage_half(Btile, kT3, 1, 1, 0);
    __builtin_amdgcn_s_barrier();
    mfma_quad(4, 0, af, b01);
    if (hT3)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else if (hT2)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    asm volatile("" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();  // staging quiet before LDS becomes the slab

  // ---- epilogue: per-wave transpose slab, one 16-row round per mi
  // fragment; wave-local lgkmcnt ordering only; each lane emits one
  // dwordx4 store per row pass (T21) -------------------------------
  {
    constexpr int SLAB_W = 68;  // 64 + 4 f32 pad
    float* slab = reinterpret_cast<float*>(smem16) + wid * 16 * SLAB_W;
    const int grow0 = tm * G16_BM + mw * 128;
    const int gcol0 = tn * G16_BN + nw * 64;
    const int c8 = lane & 7;
    const int pr = lane >> 3;

    float bv[8];
    if (HAS_BIAS) {
      const s16x8 raw =
          *reinterpret_cast<const s16x8*>(bias + gcol0 + c8 * 8);
      bf8_to_f32(raw, bv);
    }

#pragma unroll
    for (int rd = 0; rd < 8; ++rd) {  // one 16-row mi fragment each
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const f32x4 v = acc[rd][ni];
        const int col = ni * 16 + (lane & 15);
        const int row0 = (lane >> 4) * 4;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          slab[(row0 + r) * SLAB_W + col] = v[r];
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
      for (int p = 0; p < 2; ++p) {
        const int r = p * 8 + pr;
        unsigned short o[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float x = slab[r * SLAB_W + c8 * 8 + e];
          if (HAS_BIAS) x += bv[e];
          if (ACT == G16_GELU_TANH) x = fast_gelu(x);
          else if (ACT == G16_GELU_ERF) x = gelu_erf(x);
          else if (ACT == G16_SILU) x = silu(x);
          o[e] = f2bf(x);
        }
        *reinterpret_cast<i32x4*>(
            C + (long)(grow0 + rd * 16 + r) * N + gcol0 + c8 * 8) =
            *reinterpret_cast<const i32x4*>(o);
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    }
  }
}

extern "C" void launch_gemm16_bf16(const void* A, const void* B,
                                   const void* bias, void* C, int M, int N,
                                   int K, int act, hipStream_t stream) {
  const int nwg = (M / G16_BM) * (N / G16_BN);
#define G16D(ACT, HB)                                                         \
  hipLaunchKernelGGL((gemm16_kernel<ACT, HB>), dim3(nwg), dim3(512),          \
                     G16_LDS, stream, (const unsigned short*)A,               \
                     (const unsigned short*)B, (const unsigned short*)bias,   \
                     (unsigned short*)C, M, N, K)
  const bool hb = bias != nullptr;
  switch (act) {
    case G16_GELU_TANH: if (hb) G16D(1, true); else G16D(1, false); break;
    case G16_GELU_ERF: if (hb) G16D(2, true); else G16D(2, false); break;
    case G16_SILU: if (hb) G16D(3, true); else G16D(3, false); break;
    default: if (hb) G16D(0, true); else G16D(0, false); break;
  }
#undef G16D
}
