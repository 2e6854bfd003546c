// hipBLASLt epilogue-fused GEMMs for the transformer MLP hot path:
//   fwd: a = gelu(x @ Wfc^T + b1)           (GELU_AUX_BIAS, aux = pre-gelu)
//   bwd: dh1 = dgelu(dy @ Wproj, aux), db1  (DGELU_BGRAD, one GEMM)
// folding the gelu fwd kernel, gelu bwd kernel and the bias-grad column
// reduce into the GEMMs (reference keeps these as separate Gelu.cu /
// reduce kernels around cuBLAS).
//
// Row-major torch tensors are mapped to hipBLASLt's column-major world by
// computing the transposed problem (D_cm[N,M] = op(A) op(B)).
#include <hipblaslt/hipblaslt.h>
#include <torch/extension.h>

#include <mutex>
#include <unordered_map>

#include "ext_stream.h"

#define LT_CHECK(x)                                                        \
  do {                                                                     \
    hipblasStatus_t st_ = (x);                                             \
    TORCH_CHECK(st_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)st_, \
                " at " #x);                                                \
  } while (0)

namespace {

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    LT_CHECK(hipblasLtCreate(&hh));
    return hh;
  }();
  return h;
}

torch::Tensor& lt_workspace() {
  static torch::Tensor ws;
  if (!ws.defined()) {
    ws = torch::empty({64 << 20},
                      torch::dtype(torch::kUInt8).device(torch::kCUDA));
  }
  return ws;
}

struct LtPlan {
  hipblasLtMatmulDesc_t desc{};
  hipblasLtMatrixLayout_t la{}, lb{}, lc{};
  hipblasLtMatmulAlgo_t algo{};
  bool ready = false;
};

// key: (kind, M, N, K)
std::unordered_map<std::string, LtPlan> g_plans;
std::mutex g_mu;

LtPlan& get_plan(const std::string& key) { return g_plans[key]; }

void set_epilogue(hipblasLtMatmulDesc_t desc, hipblasLtEpilogue_t ep,
                  const void* bias_ptr, hipDataType bias_type,
                  const void* aux_ptr, int64_t aux_ld,
                  hipDataType aux_type) {
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      desc, HIPBLASLT_MATMUL_DESC_EPILOGUE, &ep, sizeof(ep)));
  if (bias_ptr) {
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias_ptr,
        sizeof(bias_ptr)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bias_type,
        sizeof(bias_type)));
  }
  if (aux_ptr) {
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux_ptr,
        sizeof(aux_ptr)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &aux_ld,
        sizeof(aux_ld)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE, &aux_type,
        sizeof(aux_type)));
  }
}

// col-major D[m,n] = op(A)[m,k] @ op(B)[k,n] (+ epilogue)
void lt_matmul(const std::string& key, hipblasOperation_t opa,
               hipblasOperation_t opb, int64_t m, int64_t n, int64_t k,
               const void* A, int64_t lda, const void* B, int64_t ldb,
               void* D, int64_t ldd, hipblasLtEpilogue_t ep,
               const void* bias_ptr, hipDataType bias_type,
               const void* aux_ptr, int64_t aux_ld, hipDataType aux_type) {
  std::lock_guard<std::mutex> lk(g_mu);
  auto& plan = get_plan(key);
  if (!plan.ready) {
    LT_CHECK(hipblasLtMatmulDescCreate(&plan.desc,
                                       HIPBLAS_COMPUTE_32F, HIP_R_32F));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        plan.desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opa, sizeof(opa)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        plan.desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opb, sizeof(opb)));
    LT_CHECK(hipblasLtMatrixLayoutCreate(
        &plan.la, HIP_R_16BF, opa == HIPBLAS_OP_N ? m : k,
        opa == HIPBLAS_OP_N ? k : m, lda));
    LT_CHECK(hipblasLtMatrixLayoutCreate(
        &plan.lb, HIP_R_16BF, opb == HIPBLAS_OP_N ? k : n,
        opb == HIPBLAS_OP_N ? n : k, ldb));
    LT_CHECK(hipblasLtMatrixLayoutCreate(&plan.lc, HIP_R_16BF, m, n, ldd));
  }
  // epilogue pointers change every call: set before heuristic/matmul
  set_epilogue(plan.desc, ep, bias_ptr, bias_type, aux_ptr, aux_ld,
               aux_type);
  if (!plan.ready) {
    hipblasLtMatmulPreference_t pref;
    LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws = (size_t)lt_workspace().numel();
    LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
    hipblasLtMatmulHeuristicResult_t res[4];
    int found = 0;
    LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
        lt_handle(), plan.desc, plan.la, plan.lb, plan.lc, plan.lc, pref,
        4, res, &found));
    TORCH_CHECK(found > 0, "hipblaslt: no algo for ", key);
    plan.algo = res[0].algo;
    hipblasLtMatmulPreferenceDestroy(pref);
    plan.ready = true;
  }
  float alpha = 1.f, beta = 0.f;
  LT_CHECK(hipblasLtMatmul(lt_handle(), plan.desc, &alpha, A, plan.la, B,
                           plan.lb, &beta, D, plan.lc, D, plan.lc,
                           &plan.algo, lt_workspace().data_ptr(),
                           (size_t)lt_workspace().numel(),
                           hetu_current_stream()));
}

}  // namespace

// a = gelu(x @ w^T + b), aux = x @ w^T + b (pre-gelu).
// x [M, K] row-major, w [N, K] row-major, b [N].  Returns {a, aux}.
std::vector<torch::Tensor> lt_linear_gelu_aux(torch::Tensor x,
                                              torch::Tensor w,
                                              torch::Tensor b) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous() && w.is_contiguous() &&
                  b.is_contiguous(),
              "lt_linear_gelu_aux: contiguous 2-D x");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16);
  int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && b.size(0) == N);
  auto a = torch::empty({M, N}, x.options());
  auto aux = torch::empty({M, N}, x.options());
  // col-major D[N, M] = A^T(w_cm[K,N]) @ B(x_cm[K,M])
  char key[96];
  snprintf(key, sizeof(key), "fg_%ld_%ld_%ld", (long)M, (long)N, (long)K);
  lt_matmul(key, HIPBLAS_OP_T, HIPBLAS_OP_N, N, M, K, w.data_ptr(), K,
            x.data_ptr(), K, a.data_ptr(), N,
            HIPBLASLT_EPILOGUE_GELU_AUX_BIAS, b.data_ptr(), HIP_R_16BF,
            aux.data_ptr(), N, HIP_R_16BF);
  return {a, aux};
}

// dh = dgelu(dy @ w, aux); db = colsum(dh) in the same GEMM.
// dy [M, H] row-major, w [H, F] row-major (wproj), aux [M, F].
// Returns {dh [M, F], db [F]}.
std::vector<torch::Tensor> lt_dgelu_bgrad(torch::Tensor dy, torch::Tensor w,
                                          torch::Tensor aux) {
  TORCH_CHECK(dy.dim() == 2 && dy.is_contiguous() && w.is_contiguous() &&
              aux.is_contiguous());
  int64_t M = dy.size(0), H = dy.size(1), F = w.size(1);
  TORCH_CHECK(w.size(0) == H && aux.size(0) == M && aux.size(1) == F);
  auto dh = torch::empty({M, F}, dy.options());
  auto db = torch::empty({F}, dy.options());
  // col-major D[F, M] = A(w_cm[F,H]) @ B(dy_cm[H,M])
  char key[96];
  snprintf(key, sizeof(key), "dg_%ld_%ld_%ld", (long)M, (long)F, (long)H);
  lt_matmul(key, HIPBLAS_OP_N, HIPBLAS_OP_N, F, M, H, w.data_ptr(), F,
            dy.data_ptr(), H, dh.data_ptr(), F,
            HIPBLASLT_EPILOGUE_DGELU_BGRAD, db.data_ptr(), HIP_R_16BF,
            aux.data_ptr(), F, HIP_R_16BF);
  return {dh, db};
}
