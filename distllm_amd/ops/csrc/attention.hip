// Fused flash-attention forward for CDNA4 (gfx950), bf16 I/O, fp32 accum.
//
// One kernel family covers the reference's four encoder geometries
// (SURVEY.md K1-K4/K15-prefill): bidirectional + additive key-pad mask
// (BERT/PubMedBERT, ESM2/ESMC), causal (+GQA) for Mistral/Llama prefill.
// The reference delegated all of this to HF transformers / faesm /
// flash-attn (reference: distllm/embed/encoders/*.py).
//
// v2 structure (guide Appendix B "fused attention prefill" + §5.5):
//  - workgroup = 256 threads = 4 waves; each wave owns 32 q rows
//    (128 q rows per workgroup); KV tiles of 64 keys, double-buffered,
//    ONE barrier per tile.
//  - swapped QK^T: accP = mfma_f32_32x32x16_bf16(A=K, B=Q^T) so each
//    lane holds 32 of the 64 scores of ONE q row -> online softmax is
//    lane-local + one shfl_xor(32) combine.
//  - PV is swapped too: accO^T = mfma(A=V^T, B=P): the O accumulator
//    keeps q on lanes, so the alpha rescale is lane-local.
//  - P(f32) -> bf16 B-fragments via hardware cvt + permlane32_swap.
//  - T14 async-STAGE split: next tile's V goes global->regs right at
//    the top of the iteration (HBM latency hides under QK/softmax/PV),
//    the ds_write lands after PV; K tiles stream by global_load_lds
//    (lane-linear dest, T2 XOR swizzle on the SOURCE address, rule 21).
//  - s_setprio(1) around the MFMA clusters (T5).
//  - epilogue transposes O through padded LDS for coalesced stores.
//
// Layouts: Q [B, SQ, H, D] (token stride parametric), K/V
// [B, SKV, HKV, D] (shared strides), O contiguous BSHD; mask f32
// [B, SKV] additive (0 / -inf) or nullptr.
#include "common.h"
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define KVBLK 64
#define QWAVE 32
#define LOG2E 1.4426950408889634f
#define VROW 68   // padded V^T row length in bf16 elems (136 B)

extern __shared__ __attribute__((aligned(16))) char smem[];

union frag_u {
  bf16x8 v;
  unsigned int d[4];
};

// VARLEN: Q/K/V are PACKED [total_tokens, H(KV), D] (cu_seqlens int32
// [B+1]); one workgroup per (sequence, q-block, head) via the
// host-built blk_seq/blk_qb map — the protein/fasta path with zero
// padded FLOPs (reference: the faesm flash-attn varlen role,
// distllm/embed/encoders/esm2.py:41-55).
template <int D, bool CAUSAL, bool HAS_MASK, bool VDIRECT,
          bool VARLEN = false>
__global__ void __launch_bounds__(256, 1)
attn_fwd_kernel(const unsigned short* __restrict__ Q,
                const unsigned short* __restrict__ Kt,
                const unsigned short* __restrict__ Vt,
                const float* __restrict__ maskp,
                unsigned short* __restrict__ O, int Bsz, int SQa, int SKVa,
                int H, int HKV, float scale, long bstr_q, long tstr_q,
                long bstr_k, long tstr_k,
                const int* __restrict__ cu = nullptr,
                const int* __restrict__ blk_seq = nullptr,
                const int* __restrict__ blk_qb = nullptr) {
  constexpr int NWAVE = 4;  // waves per workgroup (8 measured slower)
  constexpr int NTHR = NWAVE * 64;
  constexpr int QBLK = NWAVE * QWAVE;       // q rows per workgroup
  constexpr int NCH = D / 8;             // 16B chunks per row
  constexpr int DBLK = D / 32;           // 32-wide d blocks
  constexpr int KTILE_B = KVBLK * D * 2; // K tile bytes
  constexpr int VTILE_B = D * VROW * 2;  // V^T tile bytes
  constexpr int VCH = KVBLK * NCH / NTHR;   // V chunks per thread

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int h = lane >> 5;   // half index within wave
  const int qcol = lane & 31;

  int qb, b, head;
  if (VARLEN) {
    const int i = blockIdx.x / H;
    head = blockIdx.x % H;
    b = blk_seq[i];
    qb = blk_qb[i];
  } else {
    const int nqb0 = (SQa + QBLK - 1) / QBLK;
    // T1 bijective XCD swizzle: keep the q-blocks sharing one
    // (b, head)'s K/V on one XCD's L2 (+12% at D=128/GQA).
    int bid = blockIdx.x;
    {
      const int nwg = Bsz * H * nqb0;
      const int q = nwg / 8, r = nwg % 8;
      const int xcd = bid % 8, j = bid / 8;
      bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + j;
    }
    qb = bid % nqb0;
    const int bh = bid / nqb0;
    head = bh % H;
    b = bh / H;
  }
  const int SQ = VARLEN ? cu[b + 1] - cu[b] : SQa;
  const int SKV = VARLEN ? SQ : SKVa;
  const int kvhead = head / (H / HKV);

  const long ostride = (long)H * D;
  const long qbase = VARLEN ? (long)cu[b] * tstr_q : (long)b * bstr_q;
  const long kbase = VARLEN ? (long)cu[b] * tstr_k : (long)b * bstr_k;
  const unsigned short* Qb = Q + qbase + (long)head * D;
  const unsigned short* Kb = Kt + kbase + (long)kvhead * D;
  const unsigned short* Vb = Vt + kbase + (long)kvhead * D;
  unsigned short* Ob = O + (VARLEN ? (long)cu[b] : (long)b * SQ) * ostride +
                       (long)head * D;

  const int q0w = qb * QBLK + wid * QWAVE;
  const int qrow_abs = q0w + qcol;
  const int qload = qrow_abs < SQ ? qrow_abs : (SQ > 0 ? SQ - 1 : 0);

  const int offK[2] = {0, KTILE_B};
  const int offV[2] = {2 * KTILE_B, 2 * KTILE_B + VTILE_B};
  const int offM[2] = {2 * KTILE_B + 2 * VTILE_B,
                       2 * KTILE_B + 2 * VTILE_B + KVBLK * 4};

  const float scale2 = scale * LOG2E;

  // ---- Q fragments in registers ---------------------------------------
  bf16x8 qf[D / 16];
#pragma unroll
  for (int s = 0; s < D / 16; ++s)
    qf[s] = *reinterpret_cast<const bf16x8*>(Qb + (long)qload * tstr_q +
                                             s * 16 + h * 8);

  float m = -1e30f, l = 0.f;
  f32x16 accO[DBLK];
#pragma unroll
  for (int d = 0; d < DBLK; ++d)
#pragma unroll
    for (int i = 0; i < 16; ++i) accO[d][i] = 0.f;

  // ---- staging helpers --------------------------------------------------
  auto stage_k = [&](int k0, int buf) {   // glds, swizzled source
    constexpr int ROWS_PER = 1024 / (2 * D);   // 8 (D=64) / 4 (D=128)
    constexpr int NPIECE = KVBLK / ROWS_PER;   // 8 / 16
    const int r_in = lane / NCH;
    const int c_dst = lane % NCH;
    for (int p = wid; p < NPIECE; p += NWAVE) {
      const int row = p * ROWS_PER + r_in;
      const int c_src = c_dst ^ (row & (NCH - 1));
      int krow = k0 + row;
      if (krow >= SKV) krow = SKV - 1;
      const unsigned short* src = Kb + (long)krow * tstr_k + c_src * 8;
      auto dst = (__attribute__((address_space(3))) unsigned int*)(
          smem + offK[buf] + p * ROWS_PER * (2 * D) + lane * 16);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src, dst,
          16, 0, 0);
    }
  };
  // per-thread V staging geometry, hoisted out of the tile loop
  int v_k[VCH < 1 ? 1 : VCH];                       // key index within tile
  const unsigned short* v_src[VCH];   // &V[v_k][d0] (tile k0 = 0)
  int v_woff[VCH];                    // LDS elem offset d0*VROW + k
#pragma unroll
  for (int c = 0; c < VCH; ++c) {
    const int ci = threadIdx.x + c * NTHR;
    v_k[c] = ci / NCH;
    const int d0 = (ci % NCH) * 8;
    v_src[c] = Vb + (long)v_k[c] * tstr_k + d0;
    v_woff[c] = d0 * VROW + v_k[c];
  }
  const unsigned short* v_src_last = Vb + (long)(SKV - 1) * tstr_k;

  // T14 split: issue V loads into regs (no wait) ...
  auto v_issue = [&](int k0, s16x8* vreg) {
    const long koff = (long)k0 * tstr_k;
#pragma unroll
    for (int c = 0; c < VCH; ++c) {
      const unsigned short* src =
          (k0 + v_k[c] < SKV)
              ? v_src[c] + koff
              : v_src_last + ((v_src[c] - Vb) - (long)v_k[c] * tstr_k);
      vreg[c] = *reinterpret_cast<const s16x8*>(src);
    }
  };
  // ... and transpose-write them into the padded V^T image later.
  auto v_write = [&](int buf, const s16x8* vreg) {
    unsigned short* vt = reinterpret_cast<unsigned short*>(smem + offV[buf]);
#pragma unroll
    for (int c = 0; c < VCH; ++c) {
#pragma unroll
      for (int i = 0; i < 8; ++i)
        vt[v_woff[c] + i * VROW] = (unsigned short)vreg[c][i];
    }
  };
  auto stage_m = [&](int k0, int buf) {
    if (threadIdx.x < KVBLK) {
      const int krow = k0 + threadIdx.x;
      float mv;
      if (krow >= SKV) mv = -INFINITY;
      else if (HAS_MASK) mv = maskp[(long)b * SKV + krow] * LOG2E;
      else mv = 0.f;
      reinterpret_cast<float*>(smem + offM[buf])[threadIdx.x] = mv;
    }
  };

  int ntile = (SKV + KVBLK - 1) / KVBLK;
  if (CAUSAL) {
    const int qmax_blk = min(qb * QBLK + QBLK - 1, SQ - 1);
    ntile = min(ntile, qmax_blk / KVBLK + 1);
  }

  // prologue: stage tile 0 synchronously
  s16x8 vreg[VCH];
  stage_k(0, 0);
  if (!VDIRECT) {
    v_issue(0, vreg);
    v_write(0, vreg);
  }
  stage_m(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < ntile; ++t) {
    const int k0 = t * KVBLK;
    const bool have_next = t + 1 < ntile;
    if (have_next) {
      stage_k(k0 + KVBLK, cur ^ 1);   // async glds into the other buffer
      if (!VDIRECT) v_issue(k0 + KVBLK, vreg);  // V -> regs (T14)
    }

    const bool wave_active =
        q0w < SQ && (!CAUSAL || k0 <= q0w + QWAVE - 1);
    // mask values are only needed when a real mask exists or the tile
    // hangs over the end of the sequence (padded keys); skipping the 32
    // LDS reads otherwise removes a large per-tile VALU cost.
    const bool tile_masked = HAS_MASK || (k0 + KVBLK > SKV);

    if (wave_active) {
      // ---- QK^T: two 32-key blocks, ONE live accumulator ----------
      // (sequential kb keeps a single f32x16 accP: -16 VGPR, which is
      // what lifts the D=128 variant from 1 to 2 waves/SIMD)
      const float* mrow = reinterpret_cast<const float*>(smem + offM[cur]);
      float p[32];
      float pmax = -1e30f;
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
        f32x16 accP;
#pragma unroll
        for (int i = 0; i < 16; ++i) accP[i] = 0.f;
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int s = 0; s < D / 16; ++s) {
          const int key = kb * 32 + (lane & 31);
          const int chunk = (s * 16 + h * 8) / 8;
          const int swz = chunk ^ (key & (NCH - 1));
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              smem + offK[cur] + key * (2 * D) + swz * 16);
          accP = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              kf, qf[s], accP, 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int key_loc = kb * 32 + (r & 3) + 8 * (r >> 2) + 4 * h;
          float s2 = accP[r] * scale2;
          if (tile_masked) s2 += mrow[key_loc];
          if (CAUSAL && (k0 + key_loc > qrow_abs)) s2 = -INFINITY;
          p[kb * 16 + r] = s2;
          pmax = fmaxf(pmax, s2);
        }
      }
      pmax = fmaxf(pmax, __shfl_xor(pmax, 32, 64));
      // defer-max (guide T13, THR=8 in log2 units): skip the O rescale
      // while the running max has not grown by more than THR — P is then
      // bounded by 2^THR, which the fp32 l/O accumulation absorbs. The
      // decision covers THIS tile's P entirely (textbook-safe order:
      // nothing of this tile is exponentiated before the decision).
      float mn = m, alpha = 1.f;
      if (!__all(pmax - m <= 8.f)) {
        mn = fmaxf(fmaxf(m, pmax), -1e30f);
        alpha = __builtin_amdgcn_exp2f(m - mn);
#pragma unroll
        for (int d = 0; d < DBLK; ++d)
#pragma unroll
          for (int i = 0; i < 16; ++i) accO[d][i] *= alpha;
      }
      float rs = 0.f;
#pragma unroll
      for (int r = 0; r < 32; ++r) {
        p[r] = __builtin_amdgcn_exp2f(p[r] - mn);
        rs += p[r];
      }
      rs += __shfl_xor(rs, 32, 64);
      l = l * alpha + rs;
      m = mn;

      // ---- P -> bf16 fragments + PV, per kh (one pa live at a time:
      // -12 VGPR vs materializing all four fragments) ----------------
      const unsigned short* vimg =
          reinterpret_cast<const unsigned short*>(smem + offV[cur]);
#pragma unroll
      for (int kh = 0; kh < 4; ++kh) {
        const float* pp = p + (kh >> 1) * 16 + (kh & 1) * 8;
        unsigned int a0 = ((unsigned int)f2bf(pp[0])) |
                          ((unsigned int)f2bf(pp[1]) << 16);
        unsigned int b0 = ((unsigned int)f2bf(pp[4])) |
                          ((unsigned int)f2bf(pp[5]) << 16);
        unsigned int a1 = ((unsigned int)f2bf(pp[2])) |
                          ((unsigned int)f2bf(pp[3]) << 16);
        unsigned int b1 = ((unsigned int)f2bf(pp[6])) |
                          ((unsigned int)f2bf(pp[7]) << 16);
        i32x2 s0 = __builtin_amdgcn_permlane32_swap((int)a0, (int)b0,
                                                    false, false);
        i32x2 s1 = __builtin_amdgcn_permlane32_swap((int)a1, (int)b1,
                                                    false, false);
        frag_u pa;
        pa.d[0] = (unsigned int)s0[0];
        pa.d[1] = (unsigned int)s1[0];
        pa.d[2] = (unsigned int)s0[1];
        pa.d[3] = (unsigned int)s1[1];
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int d = 0; d < DBLK; ++d) {
          bf16x8 vf;
          if (VDIRECT) {
            // strided-but-coalesced global reads (per i: 64 lanes hit
            // 2 key rows of consecutive d) served from L1/L2
#pragma unroll
            for (int i = 0; i < 8; ++i) {
              int key = k0 + kh * 16 + h * 8 + i;
              if (key >= SKV) key = SKV - 1;
              vf[i] = *reinterpret_cast<const __bf16*>(
                  Vb + (long)key * tstr_k + d * 32 + (lane & 31));
            }
          } else {
            vf = *reinterpret_cast<const bf16x8*>(
                vimg + (d * 32 + (lane & 31)) * VROW + kh * 16 + h * 8);
          }
          accO[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              vf, pa.v, accO[d], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
    }

    if (have_next) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // V regs + glds
      if (!VDIRECT) v_write(cur ^ 1, vreg);
      stage_m(k0 + KVBLK, cur ^ 1);
    }
    __syncthreads();  // ONE barrier per tile
    cur ^= 1;
  }

  // ---- epilogue: O^T regs -> padded LDS -> coalesced bf16 stores ------
  const float inv_l = (l > 0.f) ? 1.f / l : 0.f;
  float* oimg = reinterpret_cast<float*>(smem) + wid * QWAVE * (D + 1);
#pragma unroll
  for (int d = 0; d < DBLK; ++d)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int drow = d * 32 + (r & 3) + 8 * (r >> 2) + 4 * h;
      oimg[qcol * (D + 1) + drow] = accO[d][r] * inv_l;
    }
  __syncthreads();

  if (q0w < SQ) {
    if (D == 128) {
      for (int q = 0; q < QWAVE; ++q) {
        const int qa = q0w + q;
        if (qa >= SQ) break;
        float v0 = oimg[q * (D + 1) + lane * 2];
        float v1 = oimg[q * (D + 1) + lane * 2 + 1];
        unsigned int pk = ((unsigned int)f2bf(v0)) |
                          ((unsigned int)f2bf(v1) << 16);
        *reinterpret_cast<unsigned int*>(Ob + (long)qa * ostride +
                                         lane * 2) = pk;
      }
    } else {
      for (int q = 0; q < QWAVE; q += 2) {
        const int qa = q0w + q + h;
        if (qa < SQ) {
          const int col = (lane & 31) * 2;
          float v0 = oimg[(q + h) * (D + 1) + col];
          float v1 = oimg[(q + h) * (D + 1) + col + 1];
          unsigned int pk = ((unsigned int)f2bf(v0)) |
                            ((unsigned int)f2bf(v1) << 16);
          *reinterpret_cast<unsigned int*>(Ob + (long)qa * ostride + col) =
              pk;
        }
      }
    }
  }
}

extern "C" void launch_attn_fwd(const void* Q, const void* K, const void* V,
                                const float* mask, void* O, int B, int SQ,
                                int SKV, int H, int HKV, int D, bool causal,
                                float scale, long bstr_q, long tstr_q,
                                long bstr_k, long tstr_k,
                                hipStream_t stream) {
  const int nwave = 4;
  const int qblk = nwave * 32;
  const int nqb = (SQ + qblk - 1) / qblk;
  const int grid = B * H * nqb;
  const int ktile = KVBLK * D * 2, vtile = D * VROW * 2;
  const int stage_b = 2 * ktile + 2 * vtile + 2 * KVBLK * 4;
  const int epi_b = nwave * 32 * (D + 1) * 4;
  const int lds = stage_b > epi_b ? stage_b : epi_b;

  // measured on MI355X: LDS-staged V wins at D=64 (330 vs 302 TF),
  // direct-global V wins at D=128 (176 vs 105 TF — drops the VGPR
  // pressure that pinned the kernel at 1 wave/SIMD). Env-overridable.
  static int vdirect = -1;
  if (vdirect < 0) {
    const char* e = getenv("DISTLLM_ATTN_VDIRECT");
    vdirect = e ? (e[0] == '1' ? 1 : 0) : -1;
  }
  const int vd = (vdirect >= 0) ? vdirect : (D == 128 ? 1 : 0);
#define ATTN_DISPATCH(DD, CC, MM)                                              \
  do {                                                                         \
    if (vd)                                                               \
      hipLaunchKernelGGL((attn_fwd_kernel<DD, CC, MM, true>), dim3(grid),      \
                         dim3(nwave * 64), lds, stream,                        \
                         (const unsigned short*)Q, (const unsigned short*)K,   \
                         (const unsigned short*)V, mask, (unsigned short*)O,   \
                         B, SQ, SKV, H, HKV, scale, bstr_q, tstr_q, bstr_k,    \
                         tstr_k);                                              \
    else                                                                       \
      hipLaunchKernelGGL((attn_fwd_kernel<DD, CC, MM, false>), dim3(grid),     \
                         dim3(nwave * 64), lds, stream,                        \
                         (const unsigned short*)Q, (const unsigned short*)K,   \
                         (const unsigned short*)V, mask, (unsigned short*)O,   \
                         B, SQ, SKV, H, HKV, scale, bstr_q, tstr_q, bstr_k,    \
                         tstr_k);                                              \
  } while (0)
  const bool hm = mask != nullptr;
  if (D == 64) {
    if (causal && hm) ATTN_DISPATCH(64, true, true);
    else if (causal) ATTN_DISPATCH(64, true, false);
    else if (hm) ATTN_DISPATCH(64, false, true);
    else ATTN_DISPATCH(64, false, false);
  } else if (D == 128) {
    if (causal && hm) ATTN_DISPATCH(128, true, true);
    else if (causal) ATTN_DISPATCH(128, true, false);
    else if (hm) ATTN_DISPATCH(128, false, true);
    else ATTN_DISPATCH(128, false, false);
  }
#undef ATTN_DISPATCH
}

extern "C" void launch_attn_varlen(const void* Q, const void* K,
                                   const void* V, void* O, const int* cu,
                                   const int* blk_seq, const int* blk_qb,
                                   int nblocks, int B, int H, int HKV,
                                   int D, float scale,
                                   hipStream_t stream) {
  const int nwave = 4;
  const int grid = nblocks * H;
  const int ktile = KVBLK * D * 2, vtile = D * VROW * 2;
  const int stage_b = 2 * ktile + 2 * vtile + 2 * KVBLK * 4;
  const int epi_b = nwave * 32 * (D + 1) * 4;
  const int lds = stage_b > epi_b ? stage_b : epi_b;
  const long tq = (long)H * D, tk = (long)HKV * D;
  static int vdirect = -1;
  if (vdirect < 0) {
    const char* e = getenv("DISTLLM_ATTN_VDIRECT");
    vdirect = e ? (e[0] == '1' ? 1 : 0) : -1;
  }
  const int vd = (vdirect >= 0) ? vdirect : (D == 128 ? 1 : 0);
#define AVL(DD, VD)                                                           \
  hipLaunchKernelGGL((attn_fwd_kernel<DD, false, false, VD, true>),           \
                     dim3(grid), dim3(nwave * 64), lds, stream,               \
                     (const unsigned short*)Q, (const unsigned short*)K,      \
                     (const unsigned short*)V, (const float*)nullptr,         \
                     (unsigned short*)O, B, 0, 0, H, HKV, scale, 0L, tq,      \
                     0L, tk, cu, blk_seq, blk_qb)
  if (D == 64) { if (vd) AVL(64, true); else AVL(64, false); }
  else if (D == 128) { if (vd) AVL(128, true); else AVL(128, false); }
#undef AVL
}
