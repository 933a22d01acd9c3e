// Direct hipBLASLt calls with fused epilogues (round-2 GEMM-fusion path).
//
// torch's F.linear covers the plain GEMMs (TunableOp-selected algorithms),
// but it cannot attach epilogues. These entry points fuse the MLP's
// elementwise work into the GEMMs themselves:
//   * fc1 forward:  GELU_AUX_BIAS  — bias + tanh-GELU in the GEMM epilogue,
//     pre-activation saved as the aux buffer (replaces bias_gelu_fwd_kernel).
//   * fc2 dgrad:    DGELU_BGRAD    — dGELU(aux) and the fc1 bias gradient
//     computed inside the dy@W2 GEMM (replaces bias_gelu_bwd_kernel +
//     its dbias reduction).
//
// All tensors are row-major torch tensors; hipBLASLt is column-major, so a
// row-major [M,N] output is described as a column-major [N,M] D matrix and
// the A/B operands are swapped accordingly (the standard TN mapping).
//
// Plans (descriptor + heuristic-chosen algorithm) are cached per
// (shape, epilogue); shapes repeat every training step.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>

#include <mutex>
#include <sstream>
#include <unordered_map>
#include <vector>

namespace blaslt {

#define HIPBLASLT_CHECK(expr)                                                   \
  do {                                                                          \
    hipblasStatus_t st_ = (expr);                                               \
    TORCH_CHECK(st_ == HIPBLAS_STATUS_SUCCESS, "hipBLASLt error ", (int)st_,    \
                " at " #expr);                                                  \
  } while (0)

constexpr size_t kWorkspaceBytes = 32u << 20;

hipblasLtHandle_t handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    HIPBLASLT_CHECK(hipblasLtCreate(&hh));
    return hh;
  }();
  return h;
}

void* workspace() {
  static void* ws = [] {
    void* p = nullptr;
    TORCH_CHECK(hipMalloc(&p, kWorkspaceBytes) == hipSuccess,
                "hipBLASLt workspace alloc failed");
    return p;
  }();
  return ws;
}

struct Plan {
  hipblasLtMatmulDesc_t op = nullptr;
  hipblasLtMatrixLayout_t a = nullptr, b = nullptr, c = nullptr, d = nullptr;
  hipblasLtMatmulAlgo_t algo;
};

std::unordered_map<std::string, Plan>& plan_cache() {
  static std::unordered_map<std::string, Plan> c;
  return c;
}
std::mutex& plan_mutex() {
  static std::mutex m;
  return m;
}

// Build (or fetch) a plan for D[m,n](cm) = epilogue(op(A)[m,k] * op(B)[k,n]).
// All matrix dims/lds are the column-major stored dims.
Plan& get_plan(hipblasOperation_t ta, hipblasOperation_t tb, int64_t m, int64_t n,
               int64_t k, int64_t lda, int64_t ldb, int64_t ldd,
               int64_t a_rows, int64_t a_cols, int64_t b_rows, int64_t b_cols,
               hipblasLtEpilogue_t epi, hipDataType bias_type, int64_t aux_ld) {
  std::ostringstream key;
  key << (int)ta << "," << (int)tb << "," << m << "," << n << "," << k << ","
      << lda << "," << ldb << "," << ldd << "," << (int)epi << ","
      << (int)bias_type << "," << aux_ld;
  std::lock_guard<std::mutex> lock(plan_mutex());
  auto& cache = plan_cache();
  auto it = cache.find(key.str());
  if (it != cache.end()) return it->second;

  Plan p;
  HIPBLASLT_CHECK(hipblasLtMatmulDescCreate(&p.op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_TRANSA, &ta, sizeof(ta)));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_TRANSB, &tb, sizeof(tb)));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
  if (epi != HIPBLASLT_EPILOGUE_DEFAULT) {
    int32_t bt = (int32_t)bias_type;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bt, sizeof(bt)));
  }
  if (aux_ld > 0) {
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &aux_ld, sizeof(aux_ld)));
    int32_t at = (int32_t)HIP_R_16BF;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE, &at, sizeof(at)));
  }
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.a, HIP_R_16BF, a_rows, a_cols, lda));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.b, HIP_R_16BF, b_rows, b_cols, ldb));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.c, HIP_R_16BF, m, n, ldd));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.d, HIP_R_16BF, m, n, ldd));

  hipblasLtMatmulPreference_t pref;
  HIPBLASLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  size_t ws = kWorkspaceBytes;
  HIPBLASLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
  hipblasLtMatmulHeuristicResult_t results[4];
  int n_results = 0;
  HIPBLASLT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
      handle(), p.op, p.a, p.b, p.c, p.d, pref, 4, results, &n_results));
  hipblasLtMatmulPreferenceDestroy(pref);
  TORCH_CHECK(n_results > 0, "hipBLASLt: no algorithm for epilogue ", (int)epi,
              " m=", m, " n=", n, " k=", k);
  p.algo = results[0].algo;
  return cache.emplace(key.str(), p).first->second;
}

void run(Plan& p, const void* A, const void* B, void* D, const void* bias,
         const void* aux_in, void* aux_out) {
  if (bias) {
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias, sizeof(bias)));
  }
  const void* aux = aux_out ? aux_out : aux_in;
  if (aux) {
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux, sizeof(aux)));
  }
  float alpha = 1.0f, beta = 0.0f;
  HIPBLASLT_CHECK(hipblasLtMatmul(handle(), p.op, &alpha, A, p.a, B, p.b, &beta,
                                  D, p.c, D, p.d, &p.algo, workspace(),
                                  kWorkspaceBytes, at::hip::getCurrentHIPStream().stream()));
}

void check_2d_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.dim() == 2 && t.is_contiguous() &&
                  t.scalar_type() == torch::kBFloat16,
              name, " must be contiguous 2-D bf16 on GPU");
}

// h = gelu_tanh(x @ w^T + bias), pre-activation saved as aux.
// x [M,K], w [N,K], bias [N]  ->  {h [M,N], pre [M,N]}
std::vector<torch::Tensor> gemm_bias_gelu_fwd(torch::Tensor x, torch::Tensor w,
                                              torch::Tensor bias) {
  check_2d_bf16(x, "x");
  check_2d_bf16(w, "w");
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && bias.numel() == N && bias.is_contiguous() &&
              bias.scalar_type() == torch::kBFloat16);
  auto h = torch::empty({M, N}, x.options());
  auto pre = torch::empty({M, N}, x.options());
  // D_cm[N,M] = op(A=w_cm[K,N], T)[N,K] * op(B=x_cm[K,M], N)[K,M]
  auto& p = get_plan(HIPBLAS_OP_T, HIPBLAS_OP_N, N, M, K, K, K, N,
                     K, N, K, M, HIPBLASLT_EPILOGUE_GELU_AUX_BIAS,
                     HIP_R_16BF, /*aux_ld=*/N);
  run(p, w.data_ptr(), x.data_ptr(), h.data_ptr(), bias.data_ptr(),
      nullptr, pre.data_ptr());
  return {h, pre};
}

// dpre = dgelu_tanh(dy @ w2, pre); dbias1 = column-sum of dpre (fp32).
// dy [M,N2], w2 [N2,N1], pre [M,N1]  ->  {dpre [M,N1], dbias1 [N1] fp32}
std::vector<torch::Tensor> gemm_dgelu_bgrad(torch::Tensor dy, torch::Tensor w2,
                                            torch::Tensor pre) {
  check_2d_bf16(dy, "dy");
  check_2d_bf16(w2, "w2");
  check_2d_bf16(pre, "pre");
  const int64_t M = dy.size(0), N2 = dy.size(1), N1 = w2.size(1);
  TORCH_CHECK(w2.size(0) == N2 && pre.size(0) == M && pre.size(1) == N1);
  auto dpre = torch::empty({M, N1}, dy.options());
  auto dbias = torch::empty({N1}, dy.options().dtype(torch::kFloat));
  // D_cm[N1,M] = op(A=w2_cm[N1,N2], N) * op(B=dy_cm[N2,M], N)
  auto& p = get_plan(HIPBLAS_OP_N, HIPBLAS_OP_N, N1, M, N2, N1, N2, N1,
                     N1, N2, N2, M, HIPBLASLT_EPILOGUE_DGELU_BGRAD,
                     HIP_R_32F, /*aux_ld=*/N1);
  run(p, w2.data_ptr(), dy.data_ptr(), dpre.data_ptr(), dbias.data_ptr(),
      pre.data_ptr(), nullptr);
  return {dpre, dbias};
}

// Plain bias GEMM, for validating the layout mapping: y = x @ w^T + bias.
torch::Tensor gemm_bias(torch::Tensor x, torch::Tensor w, torch::Tensor bias) {
  check_2d_bf16(x, "x");
  check_2d_bf16(w, "w");
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && bias.numel() == N);
  auto y = torch::empty({M, N}, x.options());
  auto& p = get_plan(HIPBLAS_OP_T, HIPBLAS_OP_N, N, M, K, K, K, N,
                     K, N, K, M, HIPBLASLT_EPILOGUE_BIAS, HIP_R_16BF, 0);
  run(p, w.data_ptr(), x.data_ptr(), y.data_ptr(), bias.data_ptr(), nullptr, nullptr);
  return y;
}

// Probe: how many heuristic algorithms exist for a given epilogue/dtype combo
// on this hardware (no matmul is run). aux_type/bias_type: hipDataType ints,
// -1 = leave unset. Returns n_results (0 = unsupported combo).
int probe_epilogue(int64_t epi, int64_t aux_type, int64_t bias_type,
                   int64_t m, int64_t n, int64_t k) {
  hipblasLtMatmulDesc_t op;
  if (hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F, HIP_R_32F) !=
      HIPBLAS_STATUS_SUCCESS)
    return -1;
  hipblasOperation_t ta = HIPBLAS_OP_T, tb = HIPBLAS_OP_N;
  hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSA, &ta, sizeof(ta));
  hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSB, &tb, sizeof(tb));
  hipblasLtEpilogue_t e = (hipblasLtEpilogue_t)epi;
  hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_EPILOGUE, &e, sizeof(e));
  if (bias_type >= 0) {
    int32_t bt = (int32_t)bias_type;
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bt,
                                    sizeof(bt));
  }
  if (aux_type >= 0) {
    int32_t at = (int32_t)aux_type;
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE,
                                    &at, sizeof(at));
    int64_t ld = m;
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &ld,
                                    sizeof(ld));
  }
  hipblasLtMatrixLayout_t a, b, c, d;
  hipblasLtMatrixLayoutCreate(&a, HIP_R_16BF, k, m, k);
  hipblasLtMatrixLayoutCreate(&b, HIP_R_16BF, k, n, k);
  hipblasLtMatrixLayoutCreate(&c, HIP_R_16BF, m, n, m);
  hipblasLtMatrixLayoutCreate(&d, HIP_R_16BF, m, n, m);
  hipblasLtMatmulPreference_t pref;
  hipblasLtMatmulPreferenceCreate(&pref);
  size_t ws = kWorkspaceBytes;
  hipblasLtMatmulPreferenceSetAttribute(pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES,
                                        &ws, sizeof(ws));
  hipblasLtMatmulHeuristicResult_t results[8];
  int n_results = 0;
  hipblasStatus_t st = hipblasLtMatmulAlgoGetHeuristic(handle(), op, a, b, c, d, pref,
                                                       8, results, &n_results);
  hipblasLtMatmulPreferenceDestroy(pref);
  hipblasLtMatrixLayoutDestroy(a);
  hipblasLtMatrixLayoutDestroy(b);
  hipblasLtMatrixLayoutDestroy(c);
  hipblasLtMatrixLayoutDestroy(d);
  hipblasLtMatmulDescDestroy(op);
  return st == HIPBLAS_STATUS_SUCCESS ? n_results : -(int)st;
}

}  // namespace blaslt

std::vector<torch::Tensor> blaslt_gemm_bias_gelu_fwd(torch::Tensor x, torch::Tensor w,
                                                     torch::Tensor bias) {
  return blaslt::gemm_bias_gelu_fwd(x, w, bias);
}
std::vector<torch::Tensor> blaslt_gemm_dgelu_bgrad(torch::Tensor dy, torch::Tensor w2,
                                                   torch::Tensor pre) {
  return blaslt::gemm_dgelu_bgrad(dy, w2, pre);
}
torch::Tensor blaslt_gemm_bias(torch::Tensor x, torch::Tensor w, torch::Tensor bias) {
  return blaslt::gemm_bias(x, w, bias);
}
int64_t blaslt_probe_epilogue(int64_t epi, int64_t aux_type, int64_t bias_type,
                              int64_t m, int64_t n, int64_t k) {
  return blaslt::probe_epilogue(epi, aux_type, bias_type, m, n, k);
}
