// hipBLASLt GEMM with fused BIAS+GELU epilogue (aux output = the
// pre-activation) for the GPT-2 c_fc forward: one library kernel
// replaces GEMM + separate eager GELU pass (~90 us/layer at the bench
// shape), while the aux buffer preserves the pre-activation our fused
// act_bwd kernel needs (csrc/elementwise.hip).
//
// Layout mapping (torch row-major -> hipBLASLt column-major):
//   want   out[M,N] = x[M,K] @ w[N,K]^T + bias[N]
//   compute D_cm[N,M] = op(A) * op(B) with A = w (cm [K,N], opA=T),
//   B = x (cm [K,M], opB=N) -> D_cm[N,M] IS out row-major.  The bias
//   vector (length N = rows of D_cm) is added per column = per sample,
//   matching nn.Linear.  GELU is hipBLASLt's tanh approximation — the
//   same flavor ops/linear.py uses everywhere (gelu_tanh).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>

#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

#define HIPBLASLT_CHECK(expr)                                                \
  do {                                                                       \
    hipblasStatus_t _st = (expr);                                            \
    if (_st != HIPBLAS_STATUS_SUCCESS)                                       \
      throw std::runtime_error(std::string("hipblaslt error ") +             \
                               std::to_string((int)_st) + " at " #expr);     \
  } while (0)

namespace {

constexpr size_t kWorkspaceBytes = 32ull << 20;

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    HIPBLASLT_CHECK(hipblasLtCreate(&hh));
    return hh;
  }();
  return h;
}

struct PlanKey {
  int64_t m, n, k;
  bool operator==(const PlanKey& o) const {
    return m == o.m && n == o.n && k == o.k;
  }
};
struct PlanKeyHash {
  size_t operator()(const PlanKey& p) const {
    return std::hash<int64_t>()(p.m * 1315423911 ^ p.n * 2654435761 ^ p.k);
  }
};

struct Plan {
  hipblasLtMatmulDesc_t op;
  hipblasLtMatrixLayout_t la, lb, ld;
  hipblasLtMatmulAlgo_t algo;
};

}  // namespace

// out = gelu_tanh(x @ w^T + bias); aux = x @ w^T + bias (pre-activation)
std::vector<at::Tensor> gemm_bias_gelu_aux(at::Tensor x, at::Tensor w,
                                           at::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16 && x.dim() == 2 &&
                  x.is_contiguous(),
              "gemm_bias_gelu_aux: x must be contiguous 2-D bf16 CUDA");
  TORCH_CHECK(w.is_cuda() && w.dtype() == at::kBFloat16 && w.dim() == 2 &&
                  w.is_contiguous() && w.size(1) == x.size(1),
              "gemm_bias_gelu_aux: w must be [N,K] contiguous bf16");
  TORCH_CHECK(bias.is_cuda() && bias.dtype() == at::kBFloat16 &&
                  bias.numel() == w.size(0) && bias.is_contiguous(),
              "gemm_bias_gelu_aux: bias must be [N] bf16");
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  auto out = at::empty({M, N}, x.options());
  auto aux = at::empty({M, N}, x.options());

  static std::unordered_map<PlanKey, Plan, PlanKeyHash> plans;
  PlanKey key{M, N, K};
  auto it = plans.find(key);
  if (it == plans.end()) {
    Plan p;
    HIPBLASLT_CHECK(hipblasLtMatmulDescCreate(&p.op, HIPBLAS_COMPUTE_32F,
                                              HIP_R_32F));
    int32_t opT = HIPBLAS_OP_T, opN = HIPBLAS_OP_N;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_TRANSA, &opT, sizeof(opT)));
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_TRANSB, &opN, sizeof(opN)));
    int32_t epi = HIPBLASLT_EPILOGUE_GELU_AUX_BIAS;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
    int32_t bias_t = HIP_R_16BF;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bias_t, sizeof(bias_t)));
    int64_t aux_ld = N;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &aux_ld, sizeof(aux_ld)));
    int32_t aux_t = HIP_R_16BF;
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE, &aux_t,
        sizeof(aux_t)));
    // pointers are per-call; set dummies now so the heuristic sees the
    // full epilogue configuration
    const void* dummy = out.data_ptr();
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &dummy, sizeof(dummy)));
    HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &dummy,
        sizeof(dummy)));
    // A = w: column-major [K, N]; B = x: column-major [K, M]; D: [N, M]
    HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.la, HIP_R_16BF, K, N, K));
    HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.lb, HIP_R_16BF, K, M, K));
    HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.ld, HIP_R_16BF, N, M, N));
    hipblasLtMatmulPreference_t pref;
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    uint64_t ws = kWorkspaceBytes;
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
    hipblasLtMatmulHeuristicResult_t results[8];
    int found = 0;
    HIPBLASLT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
        lt_handle(), p.op, p.la, p.lb, p.ld, p.ld, pref, 8, results, &found));
    hipblasLtMatmulPreferenceDestroy(pref);
    TORCH_CHECK(found > 0,
                "gemm_bias_gelu_aux: no hipblaslt algo for this shape");
    p.algo = results[0].algo;
    it = plans.emplace(key, p).first;
  }
  const Plan& p = it->second;

  const void* bias_ptr = bias.data_ptr();
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias_ptr, sizeof(bias_ptr)));
  const void* aux_ptr = aux.data_ptr();
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux_ptr,
      sizeof(aux_ptr)));

  auto workspace = at::empty(
      {(int64_t)kWorkspaceBytes},
      at::TensorOptions().dtype(at::kByte).device(x.device()));
  float alpha = 1.0f, beta = 0.0f;
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  HIPBLASLT_CHECK(hipblasLtMatmul(
      lt_handle(), p.op, &alpha, w.data_ptr(), p.la, x.data_ptr(), p.lb,
      &beta, out.data_ptr(), p.ld, out.data_ptr(), p.ld, &p.algo,
      workspace.data_ptr(), kWorkspaceBytes, stream));
  return {out, aux};
}
