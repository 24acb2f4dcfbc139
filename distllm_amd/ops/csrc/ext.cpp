// Torch bindings for the distllm-amd HIP kernel library (gfx950).
// Python bindings for every native op (SURVEY.md §2.9 K1-K17 kernel
// surface). The ops layer (distllm_amd/ops/__init__.py) fails loudly
// on GPU if this extension is missing — no silent eager fallback.
// All kernels are hand-written CDNA4 HIP (see *.hip in this directory);
// this file only validates tensors and forwards device pointers.
#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

#define CHECK_DEV(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_CONT(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")
#define CHECK_BF16(x) \
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, #x " must be bf16")
#define CHECK_F32(x) \
  TORCH_CHECK(x.scalar_type() == at::kFloat, #x " must be f32")
#define CHECK_I32(x) TORCH_CHECK(x.scalar_type() == at::kInt, #x " must be i32")

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

extern "C" {
void launch_norm_fwd(const void*, const void*, const void*, const void*,
                     void*, void*, int, int, float, bool, hipStream_t);
void launch_bias_gelu(const void*, const void*, void*, long, int, bool,
                      hipStream_t);
void launch_swiglu(const void*, const void*, void*, long, hipStream_t);
void launch_swiglu_packed(const void*, void*, long, int, hipStream_t);
void launch_bias_add(const void*, const void*, void*, long, int, hipStream_t);
void launch_mean_pool(const void*, const int*, float*, int, int, int, bool,
                      hipStream_t);
void launch_last_token_pool(const void*, const int*, float*, int, int, int,
                            bool, bool, hipStream_t);
void launch_l2_normalize(const float*, float*, int, int, hipStream_t);
void launch_adjacent_cosine(const float*, float*, int, int, hipStream_t);
void launch_gemm_bf16(const void*, const void*, const void*, void*, int, int,
                      int, int, hipStream_t);
void launch_gemm8_bf16(const void*, const void*, const void*, void*, int, int,
                       int, int, hipStream_t);
void launch_gemm16_bf16(const void*, const void*, const void*, void*, int,
                        int, int, int, hipStream_t);
void launch_decode_gemm(const void*, const void*, const void*, void*, void*,
                        int, int, int, int, int, hipStream_t);
void launch_attn_fwd(const void*, const void*, const void*, const float*,
                     void*, int, int, int, int, int, int, bool, float, long,
                     long, long, long, hipStream_t);
void launch_attn_varlen(const void*, const void*, const void*, void*,
                        const int*, const int*, const int*, int, int, int,
                        int, int, float, hipStream_t);
void launch_rope(const void*, void*, const float*, const float*, const int*,
                 long, int, int, int, hipStream_t);
void launch_topk_chunk(const float*, const int*, float*, int*, int, long, int,
                       int, hipStream_t);
void launch_hamming_score(const void*, const void*, float*, int, long, int,
                          hipStream_t);
void launch_ivf_score(const void*, const int*, const int*, const float*,
                      float*, int, int, int, int, hipStream_t);
void launch_gemv_fp8(const void*, const void*, const float*, const void*,
                     void*, int, int, long, hipStream_t);
void launch_sample(const float*, const float*, int*, int, int, float, float,
                   hipStream_t);
void launch_paged_attn_split(const void*, const void*, const void*,
                             const int*, const int*, void*, void*, void*,
                             int, int, int, int, int, int, float, bool,
                             hipStream_t);
void launch_paged_attn(const void*, const void*, const void*, const int*,
                       const int*, void*, int, int, int, int, int, float,
                       bool, hipStream_t);
void launch_cache_put(const void*, const void*, void*, void*, const int*,
                      const int*, int, int, int, int, bool, hipStream_t);
}

// ---- norms ----------------------------------------------------------------
torch::Tensor norm_fwd(torch::Tensor x, c10::optional<torch::Tensor> res,
                       torch::Tensor w, c10::optional<torch::Tensor> b,
                       c10::optional<torch::Tensor> sum_out, double eps,
                       bool rms) {
  CHECK_DEV(x); CHECK_CONT(x); CHECK_BF16(x);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0 && H <= 8192, "H must be mult of 8, <= 8192");
  const long n_rows = x.numel() / H;
  auto out = torch::empty_like(x);
  launch_norm_fwd(x.data_ptr(), res ? res->data_ptr() : nullptr, w.data_ptr(),
                  b ? b->data_ptr() : nullptr, out.data_ptr(),
                  sum_out ? sum_out->data_ptr() : nullptr, (int)n_rows, H,
                  (float)eps, rms, cur_stream());
  return out;
}

// ---- activations ----------------------------------------------------------
torch::Tensor bias_gelu(torch::Tensor x, c10::optional<torch::Tensor> bias,
                        bool erf) {
  CHECK_DEV(x); CHECK_CONT(x); CHECK_BF16(x);
  auto out = torch::empty_like(x);
  const int N = x.size(-1);
  TORCH_CHECK(N % 8 == 0, "last dim must be mult of 8");
  launch_bias_gelu(x.data_ptr(), bias ? bias->data_ptr() : nullptr,
                   out.data_ptr(), x.numel(), N, erf, cur_stream());
  return out;
}

torch::Tensor swiglu_packed(torch::Tensor gu) {
  CHECK_DEV(gu); CHECK_CONT(gu); CHECK_BF16(gu);
  const int I = gu.size(-1) / 2;
  TORCH_CHECK(I % 8 == 0, "I must be a multiple of 8");
  const long R = gu.numel() / (2L * I);
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto out = torch::empty(sizes, gu.options());
  launch_swiglu_packed(gu.data_ptr(), out.data_ptr(), R, I, cur_stream());
  return out;
}

torch::Tensor swiglu(torch::Tensor gate, torch::Tensor up) {
  CHECK_DEV(gate); CHECK_CONT(gate); CHECK_BF16(gate);
  auto out = torch::empty_like(gate);
  launch_swiglu(gate.data_ptr(), up.data_ptr(), out.data_ptr(), gate.numel(),
                cur_stream());
  return out;
}

torch::Tensor bias_add(torch::Tensor x, torch::Tensor bias) {
  CHECK_DEV(x); CHECK_CONT(x); CHECK_BF16(x);
  auto out = torch::empty_like(x);
  launch_bias_add(x.data_ptr(), bias.data_ptr(), out.data_ptr(), x.numel(),
                  x.size(-1), cur_stream());
  return out;
}

// ---- pooling ---------------------------------------------------------------
torch::Tensor mean_pool(torch::Tensor hidden, torch::Tensor mask,
                        bool normalize) {
  CHECK_DEV(hidden); CHECK_CONT(hidden); CHECK_BF16(hidden); CHECK_I32(mask);
  const int R = hidden.size(0), S = hidden.size(1), H = hidden.size(2);
  auto out = torch::empty({R, H}, hidden.options().dtype(at::kFloat));
  launch_mean_pool(hidden.data_ptr(), mask.data_ptr<int>(),
                   out.data_ptr<float>(), R, S, H, normalize, cur_stream());
  return out;
}

torch::Tensor last_token_pool(torch::Tensor hidden, torch::Tensor mask,
                              bool left_padded, bool normalize) {
  CHECK_DEV(hidden); CHECK_CONT(hidden); CHECK_BF16(hidden); CHECK_I32(mask);
  const int R = hidden.size(0), S = hidden.size(1), H = hidden.size(2);
  auto out = torch::empty({R, H}, hidden.options().dtype(at::kFloat));
  launch_last_token_pool(hidden.data_ptr(), mask.data_ptr<int>(),
                         out.data_ptr<float>(), R, S, H, left_padded,
                         normalize, cur_stream());
  return out;
}

torch::Tensor l2_normalize(torch::Tensor x) {
  CHECK_DEV(x); CHECK_CONT(x); CHECK_F32(x);
  const int R = x.size(0), H = x.size(1);
  auto out = torch::empty_like(x);
  launch_l2_normalize(x.data_ptr<float>(), out.data_ptr<float>(), R, H,
                      cur_stream());
  return out;
}

torch::Tensor adjacent_cosine(torch::Tensor emb) {
  CHECK_DEV(emb); CHECK_CONT(emb); CHECK_F32(emb);
  const int N = emb.size(0), H = emb.size(1);
  auto out = torch::empty({std::max(N - 1, 0)},
                          emb.options().dtype(at::kFloat));
  if (N >= 2)
    launch_adjacent_cosine(emb.data_ptr<float>(), out.data_ptr<float>(), N, H,
                           cur_stream());
  return out;
}

// ---- GEMM ------------------------------------------------------------------
torch::Tensor gemm_bf16(torch::Tensor A, torch::Tensor B,
                        c10::optional<torch::Tensor> bias, int64_t act) {
  CHECK_DEV(A); CHECK_CONT(A); CHECK_BF16(A);
  CHECK_DEV(B); CHECK_CONT(B); CHECK_BF16(B);
  const int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "K mismatch");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0,
              "gemm_bf16 requires M,N % 128 == 0 and K % 64 == 0");
  auto C = torch::empty({M, N}, A.options());
  launch_gemm_bf16(A.data_ptr(), B.data_ptr(),
                   bias ? bias->data_ptr() : nullptr, C.data_ptr(), M, N, K,
                   (int)act, cur_stream());
  return C;
}

torch::Tensor gemm8_bf16(torch::Tensor A, torch::Tensor B,
                         c10::optional<torch::Tensor> bias, int64_t act) {
  CHECK_DEV(A); CHECK_CONT(A); CHECK_BF16(A);
  CHECK_DEV(B); CHECK_CONT(B); CHECK_BF16(B);
  const int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "K mismatch");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0,
              "gemm8 requires M,N % 256 == 0 and K % 64 == 0");
  auto C = torch::empty({M, N}, A.options());
  launch_gemm8_bf16(A.data_ptr(), B.data_ptr(),
                    bias ? bias->data_ptr() : nullptr, C.data_ptr(), M, N, K,
                    (int)act, cur_stream());
  return C;
}

torch::Tensor gemm16_bf16(torch::Tensor A, torch::Tensor B,
                          c10::optional<torch::Tensor> bias, int64_t act) {
  CHECK_DEV(A); CHECK_CONT(A); CHECK_BF16(A);
  CHECK_DEV(B); CHECK_CONT(B); CHECK_BF16(B);
  const int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "K mismatch");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 128 == 0,
              "gemm16 requires M,N % 256 == 0 and K % 128 == 0");
  auto C = torch::empty({M, N}, A.options());
  launch_gemm16_bf16(A.data_ptr(), B.data_ptr(),
                     bias ? bias->data_ptr() : nullptr, C.data_ptr(), M, N,
                     K, (int)act, cur_stream());
  return C;
}

torch::Tensor decode_gemm(torch::Tensor X, torch::Tensor W,
                          c10::optional<torch::Tensor> bias,
                          int64_t splitk, int64_t act = 0) {
  // act 1 = packed swiglu in the reduce: W is [2I, K] (gate rows then
  // up rows), output is [M, I]; requires splitk > 1.
  CHECK_DEV(X); CHECK_CONT(X); CHECK_BF16(X);
  CHECK_DEV(W); CHECK_CONT(W); CHECK_BF16(W);
  const int M = X.size(0), K = X.size(1), N = W.size(0);
  TORCH_CHECK(W.size(1) == K, "K mismatch");
  TORCH_CHECK(M <= 256, "decode_gemm is for M <= 256");
  TORCH_CHECK(N % 64 == 0, "N % 64");
  TORCH_CHECK(K % (64 * splitk) == 0, "K % (64*splitk)");
  TORCH_CHECK(act == 0 || (splitk > 1 && N % 128 == 0),
              "swiglu fusion needs splitk > 1 and even I");
  const int MP = M <= 128 ? 128 : 256;
  auto out = torch::empty({M, act == 1 ? N / 2 : N}, X.options());
  torch::Tensor slabs;
  void* slabp = nullptr;
  if (splitk > 1) {
    slabs = torch::empty({splitk, (long)MP, (long)N},
                         X.options().dtype(at::kFloat));
    slabp = slabs.data_ptr();
  }
  launch_decode_gemm(X.data_ptr(), W.data_ptr(),
                     bias ? bias->data_ptr() : nullptr, out.data_ptr(),
                     slabp, M, N, K, (int)splitk, (int)act, cur_stream());
  return out;
}

// ---- attention -------------------------------------------------------------
torch::Tensor attn_fwd(torch::Tensor Q, torch::Tensor K, torch::Tensor V,
                       c10::optional<torch::Tensor> mask, bool causal,
                       double scale) {
  // Q/K/V may be strided views (e.g. slices of a packed QKV tensor) as
  // long as the innermost (D) is contiguous and heads are D-strided.
  CHECK_DEV(Q); CHECK_BF16(Q);
  CHECK_DEV(K); CHECK_BF16(K);
  CHECK_DEV(V); CHECK_BF16(V);
  const int B = Q.size(0), SQ = Q.size(1), H = Q.size(2), D = Q.size(3);
  const int SKV = K.size(1), HKV = K.size(2);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(H % HKV == 0, "GQA requires H % HKV == 0");
  TORCH_CHECK(Q.stride(3) == 1 && K.stride(3) == 1 && V.stride(3) == 1,
              "innermost dim must be contiguous");
  TORCH_CHECK(Q.stride(2) == D && K.stride(2) == D && V.stride(2) == D,
              "head stride must be D");
  TORCH_CHECK(K.stride(0) == V.stride(0) && K.stride(1) == V.stride(1),
              "K/V must share layout");
  const float* mp = nullptr;
  if (mask) {
    CHECK_DEV((*mask)); CHECK_CONT((*mask)); CHECK_F32((*mask));
    TORCH_CHECK(mask->size(0) == B && mask->size(1) == SKV,
                "mask must be [B, SKV] f32");
    mp = mask->data_ptr<float>();
  }
  auto O = torch::empty({B, SQ, H, D}, Q.options());
  launch_attn_fwd(Q.data_ptr(), K.data_ptr(), V.data_ptr(), mp, O.data_ptr(),
                  B, SQ, SKV, H, HKV, D, causal, (float)scale,
                  (long)Q.stride(0), (long)Q.stride(1), (long)K.stride(0),
                  (long)K.stride(1), cur_stream());
  return O;
}

torch::Tensor attn_varlen(torch::Tensor Q, torch::Tensor K, torch::Tensor V,
                          torch::Tensor cu, torch::Tensor blk_seq,
                          torch::Tensor blk_qb, double scale) {
  // packed [T, H(KV), D] bf16; cu int32 [B+1]; blk_* int32 [nblocks]
  CHECK_DEV(Q); CHECK_CONT(Q); CHECK_BF16(Q);
  CHECK_DEV(K); CHECK_CONT(K); CHECK_BF16(K);
  CHECK_DEV(V); CHECK_CONT(V); CHECK_BF16(V);
  CHECK_DEV(cu); CHECK_CONT(cu);
  CHECK_DEV(blk_seq); CHECK_CONT(blk_seq);
  CHECK_DEV(blk_qb); CHECK_CONT(blk_qb);
  TORCH_CHECK(cu.scalar_type() == torch::kInt32 &&
              blk_seq.scalar_type() == torch::kInt32 &&
              blk_qb.scalar_type() == torch::kInt32,
              "cu/blk maps must be int32");
  const int H = Q.size(1), D = Q.size(2), HKV = K.size(1);
  const int B = cu.size(0) - 1, nblocks = blk_seq.size(0);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(H % HKV == 0, "GQA requires H % HKV == 0");
  // zeros: rows past cu[B] (padding added to keep GEMM shapes 256-
  // aligned) are not covered by the block map and must stay inert
  auto O = torch::zeros_like(Q);
  launch_attn_varlen(Q.data_ptr(), K.data_ptr(), V.data_ptr(), O.data_ptr(),
                     cu.data_ptr<int>(), blk_seq.data_ptr<int>(),
                     blk_qb.data_ptr<int>(), nblocks, B, H, HKV, D,
                     (float)scale, cur_stream());
  return O;
}

// ---- rope ------------------------------------------------------------------
torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cs, torch::Tensor sn,
                         torch::Tensor pos, int64_t rope_heads = -1) {
  CHECK_DEV(x); CHECK_CONT(x); CHECK_BF16(x);
  CHECK_F32(cs); CHECK_F32(sn); CHECK_I32(pos);
  const int D = x.size(-1);
  const int Hx = x.size(-2);
  const long n_rows = x.numel() / ((long)D * Hx) * Hx;
  auto out = torch::empty_like(x);
  launch_rope(x.data_ptr(), out.data_ptr(), cs.data_ptr<float>(),
              sn.data_ptr<float>(), pos.data_ptr<int>(), n_rows, Hx, D,
              rope_heads < 0 ? Hx : (int)rope_heads, cur_stream());
  return out;
}

// ---- search ----------------------------------------------------------------
std::vector<torch::Tensor> topk_chunked(torch::Tensor scores,
                                        c10::optional<torch::Tensor> in_idx,
                                        int64_t k, int64_t chunk) {
  CHECK_DEV(scores); CHECK_CONT(scores); CHECK_F32(scores);
  const int Q = scores.size(0);
  const long N = scores.size(1);
  const int nchunks = (int)((N + chunk - 1) / chunk);
  auto ov = torch::empty({Q, (long)nchunks * k},
                         scores.options().dtype(at::kFloat));
  auto oi = torch::empty({Q, (long)nchunks * k},
                         scores.options().dtype(at::kInt));
  launch_topk_chunk(scores.data_ptr<float>(),
                    in_idx ? in_idx->data_ptr<int>() : nullptr,
                    ov.data_ptr<float>(), oi.data_ptr<int>(), Q, N, (int)chunk,
                    (int)k, cur_stream());
  return {ov, oi};
}

torch::Tensor hamming_score(torch::Tensor qcodes, torch::Tensor codes_t) {
  CHECK_DEV(qcodes); CHECK_CONT(qcodes);
  CHECK_DEV(codes_t); CHECK_CONT(codes_t);
  TORCH_CHECK(qcodes.scalar_type() == at::kLong || qcodes.scalar_type() == at::kUInt64);
  const int Q = qcodes.size(0), W = qcodes.size(1);
  const long N = codes_t.size(1);
  auto out = torch::empty({Q, N}, qcodes.options().dtype(at::kFloat));
  launch_hamming_score(qcodes.data_ptr(), codes_t.data_ptr(),
                       out.data_ptr<float>(), Q, N, W, cur_stream());
  return out;
}

// ---- sampling / decode -----------------------------------------------------
torch::Tensor sample_tokens(torch::Tensor logits, torch::Tensor rand,
                            double temp, double min_p) {
  CHECK_DEV(logits); CHECK_CONT(logits); CHECK_F32(logits);
  const int B = logits.size(0), V = logits.size(1);
  auto out = torch::empty({B}, logits.options().dtype(at::kInt));
  launch_sample(logits.data_ptr<float>(), rand.data_ptr<float>(),
                out.data_ptr<int>(), B, V, (float)temp, (float)min_p,
                cur_stream());
  return out;
}

torch::Tensor paged_attn(torch::Tensor q, torch::Tensor kcache,
                         torch::Tensor vcache, torch::Tensor block_table,
                         torch::Tensor ctx_len, double scale) {
  CHECK_DEV(q); CHECK_CONT(q); CHECK_BF16(q);
  const int B = q.size(0), H = q.size(1), D = q.size(2);
  const int HKV = kcache.size(1);
  const int max_blocks = block_table.size(1);
  const bool fp8 = kcache.scalar_type() == at::kByte;
  auto out = torch::empty_like(q);
  // Flash-decoding context split when B*HKV alone underfills the chip
  // (small batches): parallelize over context chunks + combine pass.
  const int GR = (HKV > 0 && H % HKV == 0) ? H / HKV : 0;
  const bool gr_ok = (GR == 1 || GR == 2 || GR == 4 || GR == 8);
  // Measured policy (profiles/RESULTS.md): split pays when B*HKV <= 128
  // AND the context is long (max_blocks >= 64 i.e. >= 1k tokens) —
  // B=1 ctx8192: 322 -> 50 us; B=32 ctx2048 regressed with split, so
  // larger batches keep the one-pass kernel.
  int NS = 1;
  if (gr_ok && B * HKV <= 128 && max_blocks >= 64)
    NS = (int)std::min<long>(std::min(512L / (B * HKV) + 1, 16L),
                             max_blocks);
  static const int ns_env = [] {  // probe override: 0=auto, 1=off, N
    const char* e = getenv("DISTLLM_PA_NSPLIT");
    return e ? atoi(e) : 0;
  }();
  if (ns_env > 0 && gr_ok)
    NS = std::min(ns_env, max_blocks);
  if (NS > 1) {
    auto f32 = q.options().dtype(at::kFloat);
    auto pacc = torch::empty({(long)B * HKV * NS * GR, D}, f32);
    auto pml = torch::empty({(long)B * HKV * NS * GR, 2}, f32);
    launch_paged_attn_split(q.data_ptr(), kcache.data_ptr(),
                            vcache.data_ptr(), block_table.data_ptr<int>(),
                            ctx_len.data_ptr<int>(), pacc.data_ptr(),
                            pml.data_ptr(), out.data_ptr(), B, H, HKV, D,
                            max_blocks, NS, (float)scale, fp8,
                            cur_stream());
    return out;
  }
  launch_paged_attn(q.data_ptr(), kcache.data_ptr(), vcache.data_ptr(),
                    block_table.data_ptr<int>(), ctx_len.data_ptr<int>(),
                    out.data_ptr(), B, H, HKV, D, max_blocks, (float)scale,
                    fp8, cur_stream());
  return out;
}

void cache_put(torch::Tensor knew, torch::Tensor vnew, torch::Tensor kcache,
               torch::Tensor vcache, torch::Tensor block_table,
               torch::Tensor slot) {
  CHECK_DEV(knew); CHECK_CONT(knew); CHECK_BF16(knew);
  const int B = knew.size(0), HKV = knew.size(1), D = knew.size(2);
  const bool fp8 = kcache.scalar_type() == at::kByte;
  launch_cache_put(knew.data_ptr(), vnew.data_ptr(), kcache.data_ptr(),
                   vcache.data_ptr(), block_table.data_ptr<int>(),
                   slot.data_ptr<int>(), B, HKV, D, block_table.size(1),
                   fp8, cur_stream());
}

// blt_gemm.cpp: hipBLASLt GEMM with fused bias/GELU/swish epilogue.
torch::Tensor blt_linear(torch::Tensor x, torch::Tensor w,
                         c10::optional<torch::Tensor> bias, int64_t epi);


torch::Tensor ivf_score(torch::Tensor slabs, torch::Tensor lens,
                        torch::Tensor probes, torch::Tensor q) {
  CHECK_DEV(slabs); CHECK_CONT(slabs); CHECK_F32(slabs);
  CHECK_DEV(lens); CHECK_CONT(lens); CHECK_I32(lens);
  CHECK_DEV(probes); CHECK_CONT(probes); CHECK_I32(probes);
  CHECK_DEV(q); CHECK_CONT(q); CHECK_F32(q);
  const int cap = slabs.size(1), H = slabs.size(2);
  const int Q = probes.size(0), nprobe = probes.size(1);
  TORCH_CHECK(H % 4 == 0, "slab dim must be padded to a multiple of 4");
  TORCH_CHECK(q.size(0) == Q && q.size(1) == H, "q shape mismatch");
  auto out = torch::empty({Q, nprobe, cap},
                          q.options().dtype(at::kFloat));
  launch_ivf_score(slabs.data_ptr(), lens.data_ptr<int>(),
                   probes.data_ptr<int>(), q.data_ptr<float>(),
                   out.data_ptr<float>(), Q * nprobe, nprobe, cap, H,
                   cur_stream());
  return out;
}

torch::Tensor gemv_fp8(torch::Tensor x, torch::Tensor wq,
                       torch::Tensor scale,
                       c10::optional<torch::Tensor> bias) {
  CHECK_DEV(x); CHECK_CONT(x); CHECK_BF16(x);
  CHECK_DEV(wq); CHECK_CONT(wq);
  TORCH_CHECK(wq.scalar_type() == at::kByte, "wq must be uint8 (fp8)");
  CHECK_DEV(scale); CHECK_CONT(scale); CHECK_F32(scale);
  const int M = x.size(0), N = wq.size(0);
  const long K = wq.size(1);
  TORCH_CHECK(x.size(1) == K, "K mismatch");
  TORCH_CHECK(M >= 1 && M <= 8, "gemv_fp8 serves M <= 8 decode rows");
  TORCH_CHECK(K % 16 == 0, "K must be a multiple of 16");
  TORCH_CHECK(scale.size(0) == N, "scale must be per output channel");
  const void* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_DEV(bias.value()); CHECK_CONT(bias.value());
    CHECK_BF16(bias.value());
    TORCH_CHECK(bias->size(0) == N, "bias shape");
    bptr = bias->data_ptr();
  }
  auto out = torch::empty({M, N}, x.options());
  launch_gemv_fp8(x.data_ptr(), wq.data_ptr(),
                  scale.data_ptr<float>(), bptr, out.data_ptr(), M, N,
                  K, cur_stream());
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("blt_linear", &blt_linear,
        "hipBLASLt linear with fused bias/GELU/swish epilogue");
  m.def("norm_fwd", &norm_fwd, "fused residual+LayerNorm/RMSNorm");
  m.def("bias_gelu", &bias_gelu);
  m.def("swiglu", &swiglu);
  m.def("swiglu_packed", &swiglu_packed);
  m.def("bias_add", &bias_add);
  m.def("mean_pool", &mean_pool);
  m.def("last_token_pool", &last_token_pool);
  m.def("l2_normalize", &l2_normalize);
  m.def("adjacent_cosine", &adjacent_cosine);
  m.def("gemm_bf16", &gemm_bf16);
  m.def("gemm8_bf16", &gemm8_bf16);
  m.def("gemm16_bf16", &gemm16_bf16);
  m.def("decode_gemm", &decode_gemm, py::arg("X"), py::arg("W"),
        py::arg("bias"), py::arg("splitk"), py::arg("act") = 0);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_varlen", &attn_varlen);
  m.def("rope_apply", &rope_apply, py::arg("x"), py::arg("cs"),
        py::arg("sn"), py::arg("pos"), py::arg("rope_heads") = -1);
  m.def("topk_chunked", &topk_chunked);
  m.def("hamming_score", &hamming_score);
  m.def("ivf_score", &ivf_score);
  m.def("gemv_fp8", &gemv_fp8, py::arg("x"), py::arg("wq"),
        py::arg("scale"), py::arg("bias") = py::none());
  m.def("sample_tokens", &sample_tokens);
  m.def("paged_attn", &paged_attn);
  m.def("cache_put", &cache_put);
}
