// hipBLASLt epilogue-fused GEMMs for the FFN hot path (gfx950, bf16).
//
//   a = GELU(x @ W1^T + b1)  — one GEMM, HIPBLASLT_EPILOGUE_GELU_BIAS
//     (inference path; see lt_linear_gelu_bias for why training cannot
//      use epilogue fusion with this hipBLASLt build)
// Row-major torch tensors are fed to the column-major hipBLASLt API with the
// usual reinterpretation X_rm[r, c] == X_cm[c, r] (no copies).
//
// Algo selection is cached per (epilogue, m, n, k); pointers are set fresh
// on each call's matmul descriptor.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hipblaslt/hipblaslt.h>

#include <chrono>
#include <map>
#include <mutex>
#include <tuple>
#include <vector>

namespace {

#define LT_CHECK(expr)                                                       \
  do {                                                                       \
    hipblasStatus_t s_ = (expr);                                             \
    TORCH_CHECK(s_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)s_,   \
                " at " #expr);                                               \
  } while (0)

constexpr size_t kWorkspace = 64u << 20;

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t x;
    TORCH_CHECK(hipblasLtCreate(&x) == HIPBLAS_STATUS_SUCCESS,
                "hipblasLtCreate failed");
    return x;
  }();
  return h;
}

using AlgoKey = std::tuple<int, long, long, long>;  // epilogue, m, n, k

hipblasLtMatmulAlgo_t cached_algo(const AlgoKey& key,
                                  hipblasLtMatmulDesc_t desc,
                                  hipblasLtMatrixLayout_t la,
                                  hipblasLtMatrixLayout_t lb,
                                  hipblasLtMatrixLayout_t lc) {
  static std::map<AlgoKey, hipblasLtMatmulAlgo_t> cache;
  static std::mutex mu;
  {
    std::lock_guard<std::mutex> g(mu);
    auto it = cache.find(key);
    if (it != cache.end()) return it->second;
  }
  hipblasLtMatmulPreference_t pref;
  LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  size_t ws = kWorkspace;
  LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
  hipblasLtMatmulHeuristicResult_t res;
  int found = 0;
  LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(lt_handle(), desc, la, lb, lc, lc,
                                           pref, 1, &res, &found));
  hipblasLtMatmulPreferenceDestroy(pref);
  TORCH_CHECK(found > 0 && res.state == HIPBLAS_STATUS_SUCCESS,
              "hipblaslt: no algo for epilogue=", std::get<0>(key),
              " m=", std::get<1>(key), " n=", std::get<2>(key),
              " k=", std::get<3>(key));
  std::lock_guard<std::mutex> g(mu);
  cache.emplace(key, res.algo);
  return res.algo;
}

hipblasLtMatrixLayout_t mk_layout(long rows, long cols, long ld) {
  hipblasLtMatrixLayout_t l;
  LT_CHECK(hipblasLtMatrixLayoutCreate(&l, HIP_R_16BF, rows, cols, ld));
  return l;
}

void set_attr(hipblasLtMatmulDesc_t d, hipblasLtMatmulDescAttributes_t a,
              const void* v, size_t sz) {
  LT_CHECK(hipblasLtMatmulDescSetAttribute(d, a, v, sz));
}

void run_matmul(hipblasLtMatmulDesc_t desc, hipblasLtMatrixLayout_t la,
                hipblasLtMatrixLayout_t lb, hipblasLtMatrixLayout_t lc,
                const void* A, const void* B, void* C,
                const hipblasLtMatmulAlgo_t& algo, torch::Tensor& ws) {
  float alpha = 1.f, beta = 0.f;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  LT_CHECK(hipblasLtMatmul(lt_handle(), desc, &alpha, A, la, B, lb, &beta, C,
                           lc, C, lc, &algo, ws.data_ptr(), kWorkspace,
                           stream));
}

torch::Tensor workspace(const torch::Tensor& like) {
  return torch::empty({(long)kWorkspace},
                      like.options().dtype(torch::kByte));
}

void check_in(const torch::Tensor& t, const char* n) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                  t.scalar_type() == torch::kBFloat16,
              n, " must be contiguous CUDA bf16");
}

}  // namespace

// a = GELU(x @ W1^T + b1) in ONE GEMM via HIPBLASLT_EPILOGUE_GELU_BIAS.
// Inference path only: this hipBLASLt build (1.2.70200) implements BIAS /
// GELU / GELU_BIAS but NOT the aux epilogues (GELU_AUX_BIAS, DGELU_BGRAD —
// probed with lt_probe_epilogue), so the pre-activation needed for backward
// is unavailable and training keeps the csrc/bias_gelu.hip kernels.
// x: [N, D], w1: [F, D] (nn.Linear layout), b1: [F] — bf16 row-major.
torch::Tensor lt_linear_gelu_bias(torch::Tensor x, torch::Tensor w1,
                                  torch::Tensor b1) {
  check_in(x, "x"); check_in(w1, "w1"); check_in(b1, "b1");
  const long N = x.size(0), D = x.size(1), F = w1.size(0);
  TORCH_CHECK(w1.size(1) == D && b1.size(0) == F, "lt_linear_gelu_bias shapes");
  auto a = torch::empty({N, F}, x.options());

  hipblasLtMatmulDesc_t desc;
  LT_CHECK(hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  int32_t opT = HIPBLAS_OP_T, opN = HIPBLAS_OP_N;
  set_attr(desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opT, sizeof(opT));
  set_attr(desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opN, sizeof(opN));
  uint32_t epi = HIPBLASLT_EPILOGUE_GELU_BIAS;
  set_attr(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi));
  const void* bias = b1.data_ptr();
  set_attr(desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias, sizeof(bias));
  int32_t bt = HIP_R_16BF;
  set_attr(desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bt, sizeof(bt));

  // col-major view: a_cm[F, N] = (w1_cm[D, F])^T @ x_cm[D, N]
  auto la = mk_layout(D, F, D);
  auto lb = mk_layout(D, N, D);
  auto lc = mk_layout(F, N, F);
  auto algo = cached_algo({(int)epi, F, N, D}, desc, la, lb, lc);
  auto ws = workspace(x);
  run_matmul(desc, la, lb, lc, w1.data_ptr(), x.data_ptr(), a.data_ptr(),
             algo, ws);
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatmulDescDestroy(desc);
  return a;
}

// Which epilogues does this hipBLASLt build actually provide kernels for?
// (aux-based epilogue variants are not guaranteed to be implemented)
bool lt_probe_epilogue(long m, long n, long k, long epi) {
  hipblasLtMatmulDesc_t desc;
  if (hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F, HIP_R_32F) !=
      HIPBLAS_STATUS_SUCCESS)
    return false;
  int32_t opT = HIPBLAS_OP_T, opN = HIPBLAS_OP_N;
  hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opT,
                                  sizeof(opT));
  hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opN,
                                  sizeof(opN));
  uint32_t e = (uint32_t)epi;
  hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE, &e,
                                  sizeof(e));
  int32_t bt = HIP_R_16BF;
  hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE,
                                  &bt, sizeof(bt));
  int32_t at = HIP_R_16BF;
  hipblasLtMatmulDescSetAttribute(
      desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE, &at, sizeof(at));
  int64_t aux_ld = m;
  hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD,
                                  &aux_ld, sizeof(aux_ld));
  hipblasLtMatrixLayout_t la = mk_layout(k, m, k), lb = mk_layout(k, n, k),
                          lc = mk_layout(m, n, m);
  hipblasLtMatmulPreference_t pref;
  hipblasLtMatmulPreferenceCreate(&pref);
  size_t ws = kWorkspace;
  hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws));
  hipblasLtMatmulHeuristicResult_t res;
  int found = 0;
  auto st = hipblasLtMatmulAlgoGetHeuristic(lt_handle(), desc, la, lb, lc,
                                            lc, pref, 1, &res, &found);
  hipblasLtMatmulPreferenceDestroy(pref);
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatmulDescDestroy(desc);
  return st == HIPBLAS_STATUS_SUCCESS && found > 0 &&
         res.state == HIPBLAS_STATUS_SUCCESS;
}

// Time the top heuristic algos for a bf16 GEMM config on device — tells us
// whether the heuristic's first choice leaves wall time on the table
// (hipBLASLt has no offline tuner in this image; TunableOp crashed).
// Returns {algo_index, ms} pairs sorted by the heuristic, timed in order.
std::vector<double> lt_bench_algos(long m, long n, long k, bool ta, bool tb,
                                   long iters) {
  auto opts = torch::TensorOptions().dtype(torch::kBFloat16)
                  .device(torch::kCUDA);
  auto A = torch::randn({ta ? k : m, ta ? m : k}, opts).contiguous();
  auto Bm = torch::randn({tb ? n : k, tb ? k : n}, opts).contiguous();
  auto C = torch::empty({m, n}, opts);  // cm [m, n] == rm [n, m]; scratch
  hipblasLtMatmulDesc_t desc;
  TORCH_CHECK(hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F,
                                        HIP_R_32F) == HIPBLAS_STATUS_SUCCESS);
  int32_t opA = ta ? HIPBLAS_OP_T : HIPBLAS_OP_N;
  int32_t opB = tb ? HIPBLAS_OP_T : HIPBLAS_OP_N;
  hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opA,
                                  sizeof(opA));
  hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opB,
                                  sizeof(opB));
  auto la = mk_layout(ta ? k : m, ta ? m : k, ta ? k : m);
  auto lb = mk_layout(tb ? n : k, tb ? k : n, tb ? n : k);
  auto lc = mk_layout(m, n, m);
  hipblasLtMatmulPreference_t pref;
  hipblasLtMatmulPreferenceCreate(&pref);
  size_t ws = kWorkspace;
  hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws));
  hipblasLtMatmulHeuristicResult_t res[16];
  int found = 0;
  auto st = hipblasLtMatmulAlgoGetHeuristic(lt_handle(), desc, la, lb, lc,
                                            lc, pref, 16, res, &found);
  hipblasLtMatmulPreferenceDestroy(pref);
  TORCH_CHECK(st == HIPBLAS_STATUS_SUCCESS && found > 0, "no algos");
  auto wsbuf = torch::empty({(long)kWorkspace},
                            opts.dtype(torch::kByte));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  float alpha = 1.f, beta = 0.f;
  std::vector<double> out;
  for (int i = 0; i < found; ++i) {
    if (res[i].state != HIPBLAS_STATUS_SUCCESS) { out.push_back(-1.0); continue; }
    auto run1 = [&] {
      return hipblasLtMatmul(lt_handle(), desc, &alpha, A.data_ptr(), la,
                             Bm.data_ptr(), lb, &beta, C.data_ptr(), lc,
                             C.data_ptr(), lc, &res[i].algo, wsbuf.data_ptr(),
                             kWorkspace, stream);
    };
    if (run1() != HIPBLAS_STATUS_SUCCESS) { out.push_back(-1.0); continue; }
    hipStreamSynchronize(stream);
    auto t0 = std::chrono::steady_clock::now();
    for (long it = 0; it < iters; ++it) run1();
    hipStreamSynchronize(stream);
    auto dt = std::chrono::duration<double, std::milli>(
                  std::chrono::steady_clock::now() - t0).count() / iters;
    out.push_back(dt);
  }
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatmulDescDestroy(desc);
  return out;
}
