// Weight-gradient GEMM that ACCUMULATES into the engine's flat fp32 buffer.
//
// dW[N,K] (+)= dy[R,N]^T @ x[R,K] as ONE hipBLASLt call with bf16 inputs,
// fp32 C=D and beta=1, where C is the parameter's slice of the flat fp32
// accumulation buffer (the reference's accum_grads assign_add,
// optimization.py:81,93). Replaces the eager-PyTorch chain
// {bf16 wgrad GEMM -> AccumulateGrad add -> K1 upcast-add} with a single
// library GEMM: fewer launches, less HBM traffic, full fp32 accumulation.
//
// Column-major mapping: row-major x[R,K] is col-major (K x R) ld=K (opA=N);
// row-major dy[R,N] is col-major (N x R) ld=N (opB=T); C (K x N) col-major
// ld=K == row-major dW[N,K] == the accum slice.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>

#include <map>
#include <vector>
#include <mutex>
#include <tuple>

#define HIPBLASLT_CHECK(expr)                                            \
  do {                                                                   \
    hipblasStatus_t _st = (expr);                                        \
    TORCH_CHECK(_st == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", _st,  \
                " at " #expr);                                           \
  } while (0)

hipblasLtHandle_t ga_lt_handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    hipblasStatus_t st = hipblasLtCreate(&hh);
    TORCH_CHECK(st == HIPBLAS_STATUS_SUCCESS, "hipblasLtCreate failed");
    return hh;
  }();
  return h;
}

void* ga_lt_workspace() {
  static void* ws = [] {
    void* p = nullptr;
    (void)hipMalloc(&p, 64ull << 20);
    return p;
  }();
  return ws;
}

namespace {

struct LtPlan {
  hipblasLtMatmulDesc_t desc;
  hipblasLtMatrixLayout_t la, lb, lc;
  std::vector<hipblasLtMatmulHeuristicResult_t> algos;
};

constexpr size_t kWorkspaceBytes = 64ull << 20;

LtPlan& plan_for(int64_t K, int64_t N, int64_t R) {
  static std::map<std::tuple<int64_t, int64_t, int64_t>, LtPlan> cache;
  static std::mutex mu;
  std::lock_guard<std::mutex> lock(mu);
  auto key = std::make_tuple(K, N, R);
  auto it = cache.find(key);
  if (it != cache.end()) return it->second;

  LtPlan p{};
  HIPBLASLT_CHECK(hipblasLtMatmulDescCreate(&p.desc, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  hipblasOperation_t opN = HIPBLAS_OP_N, opT = HIPBLAS_OP_T;
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opN, sizeof(opN)));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opT, sizeof(opT)));
  // A = x: col-major (K x R), ld K
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.la, HIP_R_16BF, K, R, K));
  // B = dy: col-major (N x R), ld N (transposed by the desc)
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.lb, HIP_R_16BF, N, R, N));
  // C = D = accum slice: col-major (K x N), ld K
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&p.lc, HIP_R_32F, K, N, K));

  hipblasLtMatmulPreference_t pref;
  HIPBLASLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  size_t ws = kWorkspaceBytes;
  HIPBLASLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
  hipblasLtMatmulHeuristicResult_t heur[16];
  int found = 0;
  hipblasStatus_t st = hipblasLtMatmulAlgoGetHeuristic(
      ga_lt_handle(), p.desc, p.la, p.lb, p.lc, p.lc, pref, 16, heur, &found);
  hipblasLtMatmulPreferenceDestroy(pref);
  TORCH_CHECK(st == HIPBLAS_STATUS_SUCCESS && found > 0, "no wgrad algo");
  p.algos.assign(heur, heur + found);
  return cache.emplace(key, p).first->second;
}

int64_t wgrad_algo_count(int64_t K, int64_t N, int64_t R) {
  return (int64_t)plan_for(K, N, R).algos.size();
}

// dW accum_slice[N*K fp32] += dy[R,N]^T @ x[R,K]
void wgrad_acc(at::Tensor x, at::Tensor dy, at::Tensor accum_slice,
               int64_t algo_idx) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16,
              "x must be contiguous bf16");
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.scalar_type() == at::kBFloat16,
              "dy must be contiguous bf16");
  TORCH_CHECK(accum_slice.scalar_type() == at::kFloat && accum_slice.is_contiguous(),
              "accum slice must be contiguous fp32");
  const int64_t K = x.size(-1);
  const int64_t N = dy.size(-1);
  const int64_t R = x.numel() / K;
  TORCH_CHECK(dy.numel() / N == R, "row count mismatch");
  TORCH_CHECK(accum_slice.numel() == N * K, "accum slice size mismatch");

  auto& p = plan_for(K, N, R);
  const float alpha = 1.f, beta = 1.f;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int idx = (int)std::min<int64_t>(std::max<int64_t>(algo_idx, 0),
                                   (int64_t)p.algos.size() - 1);
  HIPBLASLT_CHECK(hipblasLtMatmul(
      ga_lt_handle(), p.desc, &alpha, x.data_ptr(), p.la, dy.data_ptr(), p.lb,
      &beta, accum_slice.data_ptr(), p.lc, accum_slice.data_ptr(), p.lc,
      &p.algos[idx].algo, ga_lt_workspace(), kWorkspaceBytes, stream));
}

}  // namespace

void register_blas_acc(pybind11::module_& mod) {
  mod.def("wgrad_acc", &wgrad_acc,
          "accum_slice[N,K] += dy[R,N]^T @ x[R,K] (bf16 in, fp32 accumulate)",
          pybind11::arg("x"), pybind11::arg("dy"), pybind11::arg("accum_slice"),
          pybind11::arg("algo_idx") = 0);
  mod.def("wgrad_algo_count", &wgrad_algo_count);
}
