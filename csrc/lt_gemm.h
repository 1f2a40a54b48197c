// Shared hipBLASLt matmul helper for apex_amd fused GEMM extensions
// (fused_dense, mlp). Epilogue-fused GEMMs: BIAS, RELU_AUX_BIAS,
// GELU_AUX_BIAS, DGELU_BGRAD, BGRADB — the MI355X equivalents of the
// reference's cuBLASLt epilogues (csrc/fused_dense_cuda.cu:81-829).
//
// All calls take ROW-MAJOR torch tensors; this header handles the col-major
// mapping (C_rm[m,n] == C_cm[n,m]) and caches algo heuristics per shape.
#pragma once

#include "common.h"

#include <hipblaslt/hipblaslt.h>

#include <map>
#include <mutex>
#include <tuple>

#define LT_CHECK(expr)                                                          \
  do {                                                                          \
    hipblasStatus_t _s = (expr);                                                \
    TORCH_CHECK(_s == HIPBLAS_STATUS_SUCCESS, "hipBLASLt error ", (int)_s, " at ", #expr); \
  } while (0)

inline hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t handle = [] {
    hipblasLtHandle_t h;
    LT_CHECK(hipblasLtCreate(&h));
    return h;
  }();
  return handle;
}

inline hipDataType lt_dtype(at::ScalarType t) {
  switch (t) {
    case at::ScalarType::Float: return HIP_R_32F;
    case at::ScalarType::Half: return HIP_R_16F;
    case at::ScalarType::BFloat16: return HIP_R_16BF;
    default: TORCH_CHECK(false, "lt_gemm: unsupported dtype");
  }
}

constexpr size_t LT_WORKSPACE_BYTES = 64ull << 20;

struct LtAlgoKey {
  long m, n, k;
  int opA, opB, ta, tb, tc, epi;
  bool operator<(const LtAlgoKey& o) const {
    return std::tie(m, n, k, opA, opB, ta, tb, tc, epi) <
           std::tie(o.m, o.n, o.k, o.opA, o.opB, o.ta, o.tb, o.tc, o.epi);
  }
};

// col-major primitive: C[m,n] = alpha * opA(A) * opB(B) + beta * C
// aux/bias pointers per hipblasLt epilogue semantics.
// Returns false (without launching) iff no algorithm exists for the
// requested epilogue/dtype combo and `allow_fail` is set — callers fall back
// to a split epilogue (e.g. bf16 GELU_AUX_BIAS is not implemented by
// hipBLASLt 1.2).
inline bool lt_matmul_cm(hipblasOperation_t opA, hipblasOperation_t opB, long m, long n, long k,
                         const void* A, long lda, hipDataType typeA, const void* B, long ldb,
                         hipDataType typeB, void* C, long ldc, hipDataType typeC,
                         hipblasLtEpilogue_t epi, const void* bias, hipDataType bias_type,
                         void* aux, long aux_ld, hipDataType aux_type, float alpha, float beta,
                         bool allow_fail = false) {
  auto handle = lt_handle();
  auto stream = current_stream();

  hipblasLtMatmulDesc_t desc;
  LT_CHECK(hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opA, sizeof(opA)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opB, sizeof(opB)));
  if (epi != HIPBLASLT_EPILOGUE_DEFAULT) {
    LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
  }
  if (bias) {
    LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias,
                                             sizeof(bias)));
    int32_t bt = (int32_t)bias_type;
    LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bt,
                                             sizeof(bt)));
  }
  if (aux) {
    LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER,
                                             &aux, sizeof(aux)));
    int64_t ld = aux_ld;
    LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &ld,
                                             sizeof(ld)));
    int32_t at_ = (int32_t)aux_type;
    LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE,
                                             &at_, sizeof(at_)));
  }

  hipblasLtMatrixLayout_t layA, layB, layC;
  LT_CHECK(hipblasLtMatrixLayoutCreate(&layA, typeA, opA == HIPBLAS_OP_N ? m : k,
                                       opA == HIPBLAS_OP_N ? k : m, lda));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&layB, typeB, opB == HIPBLAS_OP_N ? k : n,
                                       opB == HIPBLAS_OP_N ? n : k, ldb));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&layC, typeC, m, n, ldc));

  // workspace through the torch caching allocator (stream-ordered)
  auto ws = at::empty({(long)LT_WORKSPACE_BYTES},
                      at::TensorOptions().dtype(at::kByte).device(at::kCUDA));

  hipblasLtMatmulPreference_t pref;
  LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  size_t ws_size = LT_WORKSPACE_BYTES;
  LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES,
                                                 &ws_size, sizeof(ws_size)));

  // algo heuristic cache: one query per (shape, dtype, epilogue) per process
  // (the heuristic is host-side and would otherwise stall every launch).
  // Failures are cached too — an unsupported epilogue (e.g. bf16
  // GELU_AUX_BIAS) costs a ~27 ms Tensile solution sweep per MISS, and the
  // caller's fallback path would otherwise pay it on every step.
  static std::map<LtAlgoKey, std::pair<bool, hipblasLtMatmulHeuristicResult_t>> algo_cache;
  static std::mutex algo_mu;
  LtAlgoKey key{m, n, k, (int)opA, (int)opB, (int)typeA, (int)typeB, (int)typeC, (int)epi};

  hipblasLtMatmulHeuristicResult_t heur;
  bool ok = false;
  bool cached = false;
  {
    std::lock_guard<std::mutex> lk(algo_mu);
    auto it = algo_cache.find(key);
    if (it != algo_cache.end()) {
      ok = it->second.first;
      heur = it->second.second;
      cached = true;
    }
  }
  if (!cached) {
    int nheur = 0;
    hipblasStatus_t hst = hipblasLtMatmulAlgoGetHeuristic(handle, desc, layA, layB, layC, layC,
                                                          pref, 1, &heur, &nheur);
    ok = (hst == HIPBLAS_STATUS_SUCCESS) && nheur > 0;
    std::lock_guard<std::mutex> lk(algo_mu);
    algo_cache.emplace(key, std::make_pair(ok, heur));
  }
  if (!ok && !allow_fail) {
    TORCH_CHECK(false, "hipBLASLt: no algorithm for this GEMM (m=", m, " n=", n, " k=", k,
                " epi=", (int)epi, ")");
  }
  if (ok) {
    LT_CHECK(hipblasLtMatmul(handle, desc, &alpha, A, layA, B, layB, &beta, C, layC, C, layC,
                             &heur.algo, ws.data_ptr(), ws_size, stream));
  }

  hipblasLtMatmulPreferenceDestroy(pref);
  hipblasLtMatrixLayoutDestroy(layA);
  hipblasLtMatrixLayoutDestroy(layB);
  hipblasLtMatrixLayoutDestroy(layC);
  hipblasLtMatmulDescDestroy(desc);
  return ok;
}

// ---- row-major wrappers ----

// out[m,n] = X[m,k] @ W[n,k]^T (+bias[n], + optional activation epilogue)
// aux (if used) is row-major [m, n] with ld n.
inline bool lt_linear(const at::Tensor& X, const at::Tensor& W, at::Tensor& out,
                      const at::Tensor* bias, hipblasLtEpilogue_t epi, at::Tensor* aux,
                      bool allow_fail = false) {
  const long m = X.size(0), k = X.size(1), n = W.size(0);
  return lt_matmul_cm(HIPBLAS_OP_T, HIPBLAS_OP_N, n, m, k, W.data_ptr(), k,
               lt_dtype(W.scalar_type()),
               X.data_ptr(), k, lt_dtype(X.scalar_type()), out.data_ptr(), n,
               lt_dtype(out.scalar_type()), epi, bias ? bias->data_ptr() : nullptr,
               bias ? lt_dtype(bias->scalar_type()) : HIP_R_32F,
               aux ? aux->data_ptr() : nullptr, aux ? n : 0,
               aux ? lt_dtype(aux->scalar_type()) : HIP_R_32F, 1.f, 0.f, allow_fail);
}

// dX[m,k] = dY[m,n] @ W[n,k] (+ optional DGELU/DGELU_BGRAD with aux = gelu_in
// row-major [m, k], dbias via bias pointer)
inline bool lt_linear_dgrad(const at::Tensor& dY, const at::Tensor& W, at::Tensor& dX,
                            hipblasLtEpilogue_t epi, at::Tensor* aux, at::Tensor* dbias,
                            bool allow_fail = false) {
  const long m = dY.size(0), n = dY.size(1), k = W.size(1);
  return lt_matmul_cm(HIPBLAS_OP_N, HIPBLAS_OP_N, k, m, n, W.data_ptr(), k,
               lt_dtype(W.scalar_type()),
               dY.data_ptr(), n, lt_dtype(dY.scalar_type()), dX.data_ptr(), k,
               lt_dtype(dX.scalar_type()), epi, dbias ? dbias->data_ptr() : nullptr,
               dbias ? lt_dtype(dbias->scalar_type()) : HIP_R_32F,
               aux ? aux->data_ptr() : nullptr, aux ? k : 0,
               aux ? lt_dtype(aux->scalar_type()) : HIP_R_32F, 1.f, 0.f, allow_fail);
}

// dW[n,k] (+= if beta=1) dY[m,n]^T @ X[m,k]; optional BGRADB dbias[n].
inline void lt_linear_wgrad(const at::Tensor& X, const at::Tensor& dY, at::Tensor& dW,
                            hipblasLtEpilogue_t epi, at::Tensor* dbias, float beta) {
  const long m = X.size(0), k = X.size(1), n = dY.size(1);
  lt_matmul_cm(HIPBLAS_OP_N, HIPBLAS_OP_T, k, n, m, X.data_ptr(), k, lt_dtype(X.scalar_type()),
               dY.data_ptr(), n, lt_dtype(dY.scalar_type()), dW.data_ptr(), k,
               lt_dtype(dW.scalar_type()), epi, dbias ? dbias->data_ptr() : nullptr,
               dbias ? lt_dtype(dbias->scalar_type()) : HIP_R_32F, nullptr, 0, HIP_R_32F, 1.f,
               beta);
}
