// Fused flash-attention forward for CDNA4 (gfx950), bf16 I/O, fp32 accum.
//
// One kernel family covers the reference's four encoder geometries
// (SURVEY.md K1-K4/K15-prefill): bidirectional + additive key-pad mask
// (BERT/PubMedBERT, ESM2/ESMC), causal (+GQA) for Mistral/Llama prefill.
// The reference delegated all of this to HF transformers / faesm /
// flash-attn (reference: distllm/embed/encoders/*.py).
//
// Structure (guide Appendix B "fused attention prefill" + §5.5):
//  - workgroup = 256 threads = 4 waves; each wave owns 32 q rows
//    (4x32 = 128 q rows per workgroup); KV tiles of 32 keys.
//  - swapped QK^T: accP = mfma_f32_32x32x16_bf16(A=K, B=Q^T) so each
//    lane holds 16 of the 32 scores of ONE q row -> online softmax is
//    lane-local + one shfl_xor(32) combine.
//  - PV is also swapped: accO^T = mfma(A=V^T, B=P): the O accumulator
//    keeps q on lanes, so the alpha rescale is lane-local too.
//  - P(f32) -> bf16 B-fragments via pack + __builtin_amdgcn_permlane32_swap
//    (guide T12/T21 half-exchange semantics).
//  - K tiles staged by global_load_lds (width 16) with the T2 XOR
//    swizzle on the SOURCE address (rule 21); V tiles transposed into
//    a padded [D][40/80] LDS image so PV's A-fragment is a ds_read_b128.
//  - double-buffered staging; epilogue transposes O through padded LDS
//    for coalesced bf16 stores.
//
// Layouts: Q [B, SQ, H, D], K/V [B, SKV, HKV, D], O [B, SQ, H, D], all
// bf16 contiguous; mask: f32 [B, SKV] additive (0 / -inf) or nullptr.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define KVBLK 32
#define QWAVE 32
#define QBLK 128  // 4 waves * 32
#define LOG2E 1.4426950408889634f

extern __shared__ __attribute__((aligned(16))) char smem[];

union frag_u {
  bf16x8 v;
  unsigned int d[4];
};

template <int D, bool CAUSAL, bool HAS_MASK>
__global__ void __launch_bounds__(256, 1)
attn_fwd_kernel(const unsigned short* __restrict__ Q,
                const unsigned short* __restrict__ Kt,
                const unsigned short* __restrict__ Vt,
                const float* __restrict__ maskp,
                unsigned short* __restrict__ O, int Bsz, int SQ, int SKV,
                int H, int HKV, float scale, long bstr_q, long tstr_q,
                long bstr_k, long tstr_k) {
  constexpr int NCH = D / 8;        // 16B chunks per row
  constexpr int DBLK = D / 32;      // 32-wide d blocks
  constexpr int VROW = 40;          // padded V^T row, bf16 elems (80 B)
  constexpr int KTILE_B = KVBLK * D * 2;   // K tile bytes
  constexpr int VTILE_B = D * VROW * 2;    // V^T tile bytes

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int h = lane >> 5;          // half index within wave
  const int qcol = lane & 31;       // this lane's q row (within wave tile)

  // grid: (b * H * nqb) flattened
  const int nqb = (SQ + QBLK - 1) / QBLK;
  int bid = blockIdx.x;
  const int qb = bid % nqb;
  const int bh = bid / nqb;
  const int head = bh % H;
  const int b = bh / H;
  const int kvhead = head / (H / HKV);

  const long qstride = tstr_q;             // per-token stride in Q (elems)
  const long kstride = tstr_k;             // per-token stride in K/V
  const long ostride = (long)H * D;        // O is always contiguous BSHD
  const unsigned short* Qb = Q + (long)b * bstr_q + (long)head * D;
  const unsigned short* Kb = Kt + (long)b * bstr_k + (long)kvhead * D;
  const unsigned short* Vb = Vt + (long)b * bstr_k + (long)kvhead * D;
  unsigned short* Ob = O + ((long)b * SQ) * ostride + (long)head * D;

  const int q0w = qb * QBLK + wid * QWAVE;     // wave's first q row
  const int qrow_abs = q0w + qcol;             // this lane's q row
  const bool q_valid = qrow_abs < SQ;
  const int qload = q_valid ? qrow_abs : (SQ > 0 ? SQ - 1 : 0);

  // LDS offsets
  const int offK[2] = {0, KTILE_B};
  const int offV[2] = {2 * KTILE_B, 2 * KTILE_B + VTILE_B};
  const int offM[2] = {2 * KTILE_B + 2 * VTILE_B,
                       2 * KTILE_B + 2 * VTILE_B + 128};

  const float scale2 = scale * LOG2E;

  // ---- Q fragments in registers (D/16 x bf16x8) ------------------------
  bf16x8 qf[D / 16];
#pragma unroll
  for (int s = 0; s < D / 16; ++s)
    qf[s] = *reinterpret_cast<const bf16x8*>(Qb + (long)qload * qstride +
                                             s * 16 + h * 8);

  // ---- softmax state ----------------------------------------------------
  float m = -1e30f, l = 0.f;
  f32x16 accO[DBLK];
#pragma unroll
  for (int d = 0; d < DBLK; ++d)
#pragma unroll
    for (int i = 0; i < 16; ++i) accO[d][i] = 0.f;

  // ---- staging helpers ---------------------------------------------------
  // K tile: rows = keys (clamped to SKV-1), cols = D, XOR-swizzled source.
  auto stage_k = [&](int k0, int buf) {
    // pieces of (64*16)/(2*D) rows each; piece count = KVBLK*2*D/1024
    constexpr int ROWS_PER = 1024 / (2 * D);       // 8 (D=64) or 4 (D=128)
    constexpr int NPIECE = KVBLK / ROWS_PER;       // 4 or 8
    constexpr int CH_PER_ROW = NCH;                // 16B chunks per row
    const int r_in = lane / CH_PER_ROW;
    const int c_dst = lane % CH_PER_ROW;
#pragma unroll
    for (int p = wid; p < NPIECE; p += 4) {
      const int row = p * ROWS_PER + r_in;         // key within tile
      const int c_src = c_dst ^ (row & (NCH - 1));
      int krow = k0 + row;
      if (krow >= SKV) krow = SKV - 1;
      const unsigned short* src = Kb + (long)krow * kstride + c_src * 8;
      auto dst = (__attribute__((address_space(3))) unsigned int*)(
          smem + offK[buf] + p * ROWS_PER * (2 * D) + lane * 16);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src, dst, 16,
          0, 0);
    }
  };
  // V^T tile: [D][VROW] bf16 image; transpose during staging.
  auto stage_v = [&](int k0, int buf) {
    constexpr int CH_PER_ROW = NCH;  // s16x8 chunks per source row
    const int total_chunks = KVBLK * CH_PER_ROW;   // 256 (D=64) / 512
#pragma unroll
    for (int c = threadIdx.x; c < total_chunks; c += 256) {
      const int k = c / CH_PER_ROW;
      const int d0 = (c % CH_PER_ROW) * 8;
      int krow = k0 + k;
      if (krow >= SKV) krow = SKV - 1;
      s16x8 v = *reinterpret_cast<const s16x8*>(Vb + (long)krow * kstride + d0);
      unsigned short* vt =
          reinterpret_cast<unsigned short*>(smem + offV[buf]);
#pragma unroll
      for (int i = 0; i < 8; ++i) vt[(d0 + i) * VROW + k] = (unsigned short)v[i];
    }
  };
  auto stage_m = [&](int k0, int buf) {
    if (threadIdx.x < KVBLK) {
      const int krow = k0 + threadIdx.x;
      float mv;
      if (krow >= SKV) {
        mv = -INFINITY;
      } else if (HAS_MASK) {
        mv = maskp[(long)b * SKV + krow] * LOG2E;
      } else {
        mv = 0.f;
      }
      reinterpret_cast<float*>(smem + offM[buf])[threadIdx.x] = mv;
    }
  };

  // KV loop bound: causal blocks end at the block's max q row.
  int ntile = (SKV + KVBLK - 1) / KVBLK;
  if (CAUSAL) {
    const int qmax_blk = min(qb * QBLK + QBLK - 1, SQ - 1);
    ntile = min(ntile, qmax_blk / KVBLK + 1);
  }

  stage_k(0, 0);
  stage_v(0, 0);
  stage_m(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < ntile; ++t) {
    const int k0 = t * KVBLK;
    if (t + 1 < ntile) {
      stage_k(k0 + KVBLK, cur ^ 1);
      stage_v(k0 + KVBLK, cur ^ 1);
      stage_m(k0 + KVBLK, cur ^ 1);
    }

    // wave-level skip: causal tiles fully after this wave's rows
    const bool wave_active =
        q0w < SQ && (!CAUSAL || k0 <= q0w + QWAVE - 1);

    if (wave_active) {
      // ---- QK^T: accP[key16][q] ---------------------------------------
      f32x16 accP;
#pragma unroll
      for (int i = 0; i < 16; ++i) accP[i] = 0.f;
#pragma unroll
      for (int s = 0; s < D / 16; ++s) {
        // A-frag: K[key = lane&31][d = s*16 + h*8 + 0..7]
        const int key = lane & 31;
        const int chunk = (s * 16 + h * 8) / 8;
        const int swz = chunk ^ (key & (NCH - 1));
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            smem + offK[cur] + key * (2 * D) + swz * 16);
        accP = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], accP, 0, 0,
                                                       0);
      }

      // ---- mask + causal + online softmax ------------------------------
      const float* mrow = reinterpret_cast<const float*>(smem + offM[cur]);
      float p[16];
      float pmax = -1e30f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int key_loc = (r & 3) + 8 * (r >> 2) + 4 * h;
        float s2 = accP[r] * scale2 + mrow[key_loc];
        if (CAUSAL && (k0 + key_loc > qrow_abs)) s2 = -INFINITY;
        p[r] = s2;
        pmax = fmaxf(pmax, s2);
      }
      pmax = fmaxf(pmax, __shfl_xor(pmax, 32, 64));
      const float mn = fmaxf(fmaxf(m, pmax), -1e30f);
      const float alpha = exp2f(m - mn);
      float rs = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = exp2f(p[r] - mn);
        rs += p[r];
      }
      rs += __shfl_xor(rs, 32, 64);
      l = l * alpha + rs;
      m = mn;
#pragma unroll
      for (int d = 0; d < DBLK; ++d)
#pragma unroll
        for (int i = 0; i < 16; ++i) accO[d][i] *= alpha;

      // ---- P -> bf16 B-fragments (pack + permlane32_swap) --------------
      // pair (w0,w2) = swap(pack(r0,r1), pack(r4,r5)) etc.  (see header)
      frag_u pa[2];
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
        unsigned int a0 = ((unsigned int)f2bf(p[kh * 8 + 0])) |
                          ((unsigned int)f2bf(p[kh * 8 + 1]) << 16);
        unsigned int b0 = ((unsigned int)f2bf(p[kh * 8 + 4])) |
                          ((unsigned int)f2bf(p[kh * 8 + 5]) << 16);
        unsigned int a1 = ((unsigned int)f2bf(p[kh * 8 + 2])) |
                          ((unsigned int)f2bf(p[kh * 8 + 3]) << 16);
        unsigned int b1 = ((unsigned int)f2bf(p[kh * 8 + 6])) |
                          ((unsigned int)f2bf(p[kh * 8 + 7]) << 16);
        i32x2 s0 = __builtin_amdgcn_permlane32_swap((int)a0, (int)b0, false,
                                                    false);
        i32x2 s1 = __builtin_amdgcn_permlane32_swap((int)a1, (int)b1, false,
                                                    false);
        pa[kh].d[0] = (unsigned int)s0[0];
        pa[kh].d[1] = (unsigned int)s1[0];
        pa[kh].d[2] = (unsigned int)s0[1];
        pa[kh].d[3] = (unsigned int)s1[1];
      }

      // ---- PV: accO^T[d16][q] += V^T frag x P frag ---------------------
      const unsigned short* vimg =
          reinterpret_cast<const unsigned short*>(smem + offV[cur]);
#pragma unroll
      for (int d = 0; d < DBLK; ++d) {
#pragma unroll
        for (int kh = 0; kh < 2; ++kh) {
          // A-frag: V^T[row = d*32 + (lane&31)][k = kh*16 + h*8 + 0..7]
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(
              vimg + (d * 32 + (lane & 31)) * VROW + kh * 16 + h * 8);
          accO[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pa[kh].v,
                                                            accO[d], 0, 0, 0);
        }
      }
    }

    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: O^T regs -> padded LDS -> coalesced bf16 stores -------
  const float inv_l = (l > 0.f) ? 1.f / l : 0.f;
  float* oimg = reinterpret_cast<float*>(smem) + wid * QWAVE * (D + 1);
#pragma unroll
  for (int d = 0; d < DBLK; ++d)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int drow = d * 32 + (r & 3) + 8 * (r >> 2) + 4 * h;
      oimg[qcol * (D + 1) + drow] = accO[d][r] * inv_l;
    }
  __syncthreads();  // all waves done writing (regions disjoint; keeps
                    // the staging buffers from being clobbered early)

  // write: each iteration covers 64/ (D/32... ) use: lane handles 2 cols
  // per q row for D=128, 1 col-pair rows for D=64.
  if (q0w < SQ) {
    if (D == 128) {
      for (int q = 0; q < QWAVE; ++q) {
        const int qa = q0w + q;
        if (qa >= SQ) break;
        float v0 = oimg[q * (D + 1) + lane * 2];
        float v1 = oimg[q * (D + 1) + lane * 2 + 1];
        unsigned int pk = ((unsigned int)f2bf(v0)) |
                          ((unsigned int)f2bf(v1) << 16);
        *reinterpret_cast<unsigned int*>(Ob + (long)qa * ostride + lane * 2) =
            pk;
      }
    } else {
      // D == 64: two q rows per iteration
      for (int q = 0; q < QWAVE; q += 2) {
        const int qa = q0w + q + h;
        if (qa < SQ) {
          const int col = (lane & 31) * 2;
          float v0 = oimg[(q + h) * (D + 1) + col];
          float v1 = oimg[(q + h) * (D + 1) + col + 1];
          unsigned int pk = ((unsigned int)f2bf(v0)) |
                            ((unsigned int)f2bf(v1) << 16);
          *reinterpret_cast<unsigned int*>(Ob + (long)qa * ostride + col) = pk;
        }
      }
    }
  }
}

extern "C" void launch_attn_fwd(const void* Q, const void* K, const void* V,
                                const float* mask, void* O, int B, int SQ,
                                int SKV, int H, int HKV, int D, bool causal,
                                float scale, long bstr_q, long tstr_q,
                                long bstr_k, long tstr_k, hipStream_t stream) {
  const int nqb = (SQ + QBLK - 1) / QBLK;
  const int grid = B * H * nqb;
  // LDS: staging (2K + 2V + 2mask) or epilogue (4 waves * 32 * (D+1) f32)
  const int ktile = KVBLK * D * 2, vtile = D * 40 * 2;
  const int stage_b = 2 * ktile + 2 * vtile + 256;
  const int epi_b = 4 * QWAVE * (D + 1) * 4;
  const int lds = stage_b > epi_b ? stage_b : epi_b;

#define ATTN_DISPATCH(DD, CC, MM)                                              \
  hipLaunchKernelGGL((attn_fwd_kernel<DD, CC, MM>), dim3(grid), dim3(256),     \
                     lds, stream, (const unsigned short*)Q,                    \
                     (const unsigned short*)K, (const unsigned short*)V, mask, \
                     (unsigned short*)O, B, SQ, SKV, H, HKV, scale, bstr_q,    \
                     tstr_q, bstr_k, tstr_k)
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
