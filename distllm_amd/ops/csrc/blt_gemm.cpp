// hipBLASLt linear with a FUSED epilogue (bias / bias+GELU / bias+swish).
//
// The reference runs HF-BERT's FFN as GEMM -> separate bias+GELU pass
// (reference: distllm/embed/encoders/*, via HF transformers). On MI355X
// the separate pass re-reads and re-writes the whole [M, 4H] bf16
// intermediate over HBM3E (~3.2 GB per FFN1 at bs512xseq512) — ~12% of
// the flagship embed step. hipBLASLt applies the activation inside the
// GEMM epilogue while the tile is still in registers, so that traffic
// disappears entirely. The Python op layer autotunes this path against
// (hipBLASLt + standalone fused kernel) and gemm8 per shape.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>

#include <map>
#include <tuple>

#define BLT_CHECK(x)                                                        \
  do {                                                                      \
    hipblasStatus_t st_ = (x);                                              \
    TORCH_CHECK(st_ == HIPBLAS_STATUS_SUCCESS, "hipBLASLt error ",          \
                (int)st_, " at " #x);                                       \
  } while (0)

// epi: 0 = none/bias-only, 1 = GELU(+bias), 2 = swish/silu(+bias)
torch::Tensor blt_linear(torch::Tensor x, torch::Tensor w,
                         c10::optional<torch::Tensor> bias, int64_t epi) {
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1),
              "blt_linear: shape mismatch");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16,
              "blt_linear: bf16 only");
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  auto out = torch::empty({M, N}, x.options());

  static hipblasLtHandle_t handle = nullptr;
  if (!handle) BLT_CHECK(hipblasLtCreate(&handle));

  hipblasLtMatmulDesc_t op;
  BLT_CHECK(hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  const hipblasOperation_t ta = HIPBLAS_OP_T, tb = HIPBLAS_OP_N;
  BLT_CHECK(hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSA,
                                            &ta, sizeof(ta)));
  BLT_CHECK(hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSB,
                                            &tb, sizeof(tb)));
  const bool has_bias = bias.has_value();
  hipblasLtEpilogue_t ep;
  if (epi == 1)
    ep = has_bias ? HIPBLASLT_EPILOGUE_GELU_BIAS : HIPBLASLT_EPILOGUE_GELU;
  else if (epi == 2)
    ep = has_bias ? HIPBLASLT_EPILOGUE_SWISH_BIAS_EXT
                  : HIPBLASLT_EPILOGUE_SWISH_EXT;
  else
    ep = has_bias ? HIPBLASLT_EPILOGUE_BIAS : HIPBLASLT_EPILOGUE_DEFAULT;
  BLT_CHECK(hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_EPILOGUE,
                                            &ep, sizeof(ep)));
  if (has_bias) {
    TORCH_CHECK(bias->scalar_type() == at::kBFloat16 && bias->numel() == N);
    const void* bp = bias->data_ptr();
    BLT_CHECK(hipblasLtMatmulDescSetAttribute(
        op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bp, sizeof(bp)));
    int32_t bt = (int32_t)HIP_R_16BF;
    BLT_CHECK(hipblasLtMatmulDescSetAttribute(
        op, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bt, sizeof(bt)));
  }
  // Column-major mapping of the row-major torch call
  //   out[M,N] = x[M,K] @ W[N,K]^T :
  // D_col[N,M] = op(A) op(B), A = W stored col-major [K,N] (opA = T),
  // B = x stored col-major [K,M] (opB = N), ldd = N. The bias vector
  // length equals D's rows = N, i.e. one value per output feature.
  hipblasLtMatrixLayout_t la, lb, lc;
  BLT_CHECK(hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, K, N, K));
  BLT_CHECK(hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, K, M, K));
  BLT_CHECK(hipblasLtMatrixLayoutCreate(&lc, HIP_R_16BF, N, M, N));

  // One workspace + one algo (heuristic #1) per (shape, epilogue).
  constexpr size_t WS_BYTES = size_t(64) << 20;
  auto ws = torch::empty({(int64_t)WS_BYTES}, x.options().dtype(at::kByte));
  using Key = std::tuple<int64_t, int64_t, int64_t, int, bool>;
  static std::map<Key, hipblasLtMatmulAlgo_t> algo_cache;
  const Key key{M, N, K, (int)ep, has_bias};
  auto it = algo_cache.find(key);
  if (it == algo_cache.end()) {
    hipblasLtMatmulPreference_t pref;
    BLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    uint64_t wsz = WS_BYTES;
    BLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &wsz, sizeof(wsz)));
    hipblasLtMatmulHeuristicResult_t heur[16];
    int nret = 0;
    BLT_CHECK(hipblasLtMatmulAlgoGetHeuristic(handle, op, la, lb, lc, lc,
                                              pref, 16, heur, &nret));
    hipblasLtMatmulPreferenceDestroy(pref);
    TORCH_CHECK(nret > 0, "hipBLASLt: no algo for fused epilogue M=", M,
                " N=", N, " K=", K, " ep=", (int)ep);
    // Heuristic order is an estimate; measure the candidates (first
    // call per shape only). In-order timing is biased — early
    // candidates run on cold clocks — so warm the GPU first and only
    // leave heuristic #0 for a candidate that beats it by > 3%.
    int best = 0;
    if (nret > 1) {
      auto stream = c10::hip::getCurrentHIPStream().stream();
      const float alpha_ = 1.f, beta_ = 0.f;
      auto run = [&](int i) {
        return hipblasLtMatmul(handle, op, &alpha_, w.data_ptr(), la,
                               x.data_ptr(), lb, &beta_, out.data_ptr(),
                               lc, out.data_ptr(), lc, &heur[i].algo,
                               ws.data_ptr(), WS_BYTES, stream);
      };
      hipEvent_t e0, e1;
      (void)hipEventCreate(&e0);
      (void)hipEventCreate(&e1);
      for (int r = 0; r < 6; ++r) (void)run(0);  // clock warmup
      float ms0 = 1e30f, best_ms = 1e30f;
      for (int i = 0; i < nret; ++i) {
        if (heur[i].state != HIPBLAS_STATUS_SUCCESS) continue;
        if (run(i) != HIPBLAS_STATUS_SUCCESS) continue;  // validity
        (void)hipEventRecord(e0, stream);
        for (int r = 0; r < 4; ++r) (void)run(i);
        (void)hipEventRecord(e1, stream);
        (void)hipEventSynchronize(e1);
        float ms = 1e30f;
        (void)hipEventElapsedTime(&ms, e0, e1);
        if (i == 0) ms0 = ms;
        if (ms < best_ms) { best_ms = ms; best = i; }
      }
      if (best != 0 && best_ms > 0.97f * ms0) best = 0;
      (void)hipEventDestroy(e0);
      (void)hipEventDestroy(e1);
    }
    it = algo_cache.emplace(key, heur[best].algo).first;
  }
  const float alpha = 1.f, beta = 0.f;
  BLT_CHECK(hipblasLtMatmul(handle, op, &alpha, w.data_ptr(), la,
                            x.data_ptr(), lb, &beta, out.data_ptr(), lc,
                            out.data_ptr(), lc, &it->second, ws.data_ptr(),
                            WS_BYTES, c10::hip::getCurrentHIPStream()));
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatmulDescDestroy(op);
  return out;
}
