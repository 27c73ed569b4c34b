// Grouped weight-gradient GEMM: all of a micro-step's independent wgrads as
// ONE hipBLASLt grouped-gemm launch, each problem accumulating (beta=1,
// fp32) into its parameter's slice of the flat accum buffer.
//
// Why: at the reference's micro-batch the per-layer wgrad GEMMs are ~1-2
// GFLOP each and fill only a fraction of 256 CUs; 16 of them per micro-step
// cost ~16 launches x ~12 us. Grouped, the whole set fills the chip once.
//
// The kernel arguments (pointers incl.) are frozen at initialize() time, so
// instances are cached keyed by the full (shapes + pointers) signature --
// under hipGraph capture the activation/grad buffers are pool-stable, so
// each captured graph resolves to one cached instance and replays cleanly.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>
#include <hipblaslt/hipblaslt-ext.hpp>

#include <map>
#include <memory>
#include <vector>

hipblasLtHandle_t ga_lt_handle();  // blas_acc.hip

namespace {

constexpr size_t kGroupedWorkspace = 128ull << 20;

void* grouped_workspace() {
  static void* ws = [] {
    void* p = nullptr;
    (void)hipMalloc(&p, kGroupedWorkspace);
    return p;
  }();
  return ws;
}

static const float kOne = 1.f;

void grouped_wgrad_acc(std::vector<at::Tensor> xs, std::vector<at::Tensor> dys,
                       std::vector<at::Tensor> accs) {
  const size_t G = xs.size();
  TORCH_CHECK(G > 0 && dys.size() == G && accs.size() == G, "group size mismatch");

  // cache key: shapes + pointers (pointers are baked into kernel args)
  std::vector<int64_t> key;
  key.reserve(G * 6);
  std::vector<int64_t> m(G), n(G), k(G), batch(G, 1);
  std::vector<int64_t> lda(G), ldb(G), ldc(G), ldd(G);
  std::vector<int64_t> sA(G, 0), sB(G, 0), sC(G, 0), sD(G, 0);
  for (size_t i = 0; i < G; ++i) {
    auto& x = xs[i];
    auto& dy = dys[i];
    TORCH_CHECK(x.is_contiguous() && x.scalar_type() == at::kBFloat16, "x bf16");
    TORCH_CHECK(dy.is_contiguous() && dy.scalar_type() == at::kBFloat16, "dy bf16");
    TORCH_CHECK(accs[i].is_contiguous() && accs[i].scalar_type() == at::kFloat);
    const int64_t K = x.size(-1), N = dy.size(-1), R = x.numel() / K;
    TORCH_CHECK(dy.numel() / N == R && accs[i].numel() == N * K, "wgrad shape");
    // D_cm(K,N) = x_cm(K,R) opN @ dy_cm(N,R) opT
    m[i] = K; n[i] = N; k[i] = R;
    lda[i] = K; ldb[i] = N; ldc[i] = ldd[i] = K;
    key.push_back((int64_t)x.data_ptr());
    key.push_back((int64_t)dy.data_ptr());
    key.push_back((int64_t)accs[i].data_ptr());
    key.push_back(K); key.push_back(N); key.push_back(R);
  }

  using GG = hipblaslt_ext::GroupedGemm;
  static std::map<std::vector<int64_t>, std::unique_ptr<GG>> cache;
  auto it = cache.find(key);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (it == cache.end()) {
    auto gg = std::make_unique<GG>(
        ga_lt_handle(), HIPBLAS_OP_N, HIPBLAS_OP_T, HIP_R_16BF, HIP_R_16BF,
        HIP_R_32F, HIP_R_32F, HIPBLAS_COMPUTE_32F);
    hipblaslt_ext::GemmProblemType ptype(HIPBLAS_OP_N, HIPBLAS_OP_T, HIP_R_16BF,
                                         HIP_R_16BF, HIP_R_32F, HIP_R_32F,
                                         HIPBLAS_COMPUTE_32F);
    std::vector<hipblaslt_ext::GemmEpilogue> epi(G);
    std::vector<hipblaslt_ext::GemmInputs> inp(G);
    for (size_t i = 0; i < G; ++i) {
      inp[i].setA(xs[i].data_ptr());
      inp[i].setB(dys[i].data_ptr());
      inp[i].setC(accs[i].data_ptr());
      inp[i].setD(accs[i].data_ptr());
      inp[i].setAlpha(&kOne);
      inp[i].setBeta(&kOne);
    }
    TORCH_CHECK(gg->setProblem(m, n, k, batch, lda, ldb, ldc, ldd, sA, sB, sC,
                               sD, epi, inp, ptype) == HIPBLAS_STATUS_SUCCESS,
                "grouped wgrad setProblem failed");
    hipblaslt_ext::GemmPreference pref;
    pref.setMaxWorkspaceBytes(kGroupedWorkspace);
    std::vector<hipblasLtMatmulHeuristicResult_t> heur;
    TORCH_CHECK(gg->algoGetHeuristic(8, pref, heur) == HIPBLAS_STATUS_SUCCESS &&
                    !heur.empty(),
                "grouped wgrad: no heuristic");
    // isAlgoSupported asserts "hardware != nullptr" inside this hipblaslt
    // build for grouped problems; the heuristic list already respected the
    // workspace preference, so initialize straight from the first result.
    TORCH_CHECK(gg->initialize(heur[0].algo, grouped_workspace(), true, stream) ==
                    HIPBLAS_STATUS_SUCCESS,
                "grouped wgrad initialize failed");
    it = cache.emplace(std::move(key), std::move(gg)).first;
    // bound the cache (eager fallback churns pointer sets)
    if (cache.size() > 32) cache.erase(cache.begin());
  }
  TORCH_CHECK(it->second->run(stream) == HIPBLAS_STATUS_SUCCESS,
              "grouped wgrad run failed");
}

}  // namespace

void register_grouped_wgrad(pybind11::module_& mod) {
  mod.def("grouped_wgrad_acc", &grouped_wgrad_acc,
          "one grouped hipBLASLt launch: accum_i += dy_i^T @ x_i (fp32)");
}
