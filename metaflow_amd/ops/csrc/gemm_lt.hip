// hipBLASLt algorithm search + pinned-algo GEMM for the Llama hot shapes.
//
// The train-step profile (profiles/llama8b_1gpu_r01_final2_kernel_stats.txt)
// shows hipBLASLt's heuristic picking MI16x16 tiles for ~58% of step time.
// This extension enumerates EVERY solution hipBLASLt ships for a given
// (M, N, K) bf16 TN GEMM (hipblaslt_ext::getAllAlgos), times each on the
// real shape, and replays the winner by solution index — an offline
// tuning pass without TunableOp's in-process fragility (NOTES_ROUND2.md).
//
// Layout convention: torch F.linear semantics out[M,N] = x[M,K] @ w[N,K]^T,
// all row-major.  hipBLASLt is column-major, so we compute
// D_cm[N,M] = op_T(W_cm[K,N]) * op_N(X_cm[K,M])  — the TN GEMM the
// Cijk_Alik_Bljk solutions in the profile implement.

#include <hipblaslt/hipblaslt.h>
#include <hipblaslt/hipblaslt-ext.hpp>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

#define LT_CHECK(expr)                                                     \
  do {                                                                     \
    hipblasStatus_t s_ = (expr);                                           \
    TORCH_CHECK(s_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)s_, \
                " at " #expr);                                             \
  } while (0)

#define HIPRT_CHECK(expr)                                                 \
  do {                                                                    \
    hipError_t e_ = (expr);                                               \
    TORCH_CHECK(e_ == hipSuccess, "hip error: ", hipGetErrorString(e_));  \
  } while (0)

namespace {

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t handle = [] {
    hipblasLtHandle_t h;
    LT_CHECK(hipblasLtCreate(&h));
    return h;
  }();
  return handle;
}

struct Layouts {
  hipblasLtMatmulDesc_t op;
  hipblasLtMatrixLayout_t a, b, c;
  ~Layouts() {
    hipblasLtMatrixLayoutDestroy(a);
    hipblasLtMatrixLayoutDestroy(b);
    hipblasLtMatrixLayoutDestroy(c);
    hipblasLtMatmulDescDestroy(op);
  }
};

// D[N,M]cm = W^T[N,K] * X[K,M]cm : A = W (K x N cm, OP_T), B = X (K x M cm)
void make_layouts(Layouts& L, long long M, long long N, long long K) {
  LT_CHECK(hipblasLtMatmulDescCreate(&L.op, HIPBLAS_COMPUTE_32F,
                                     HIP_R_32F));
  hipblasOperation_t t = HIPBLAS_OP_T, n = HIPBLAS_OP_N;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      L.op, HIPBLASLT_MATMUL_DESC_TRANSA, &t, sizeof(t)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      L.op, HIPBLASLT_MATMUL_DESC_TRANSB, &n, sizeof(n)));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&L.a, HIP_R_16BF, K, N, K));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&L.b, HIP_R_16BF, K, M, K));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&L.c, HIP_R_16BF, N, M, N));
}

void run_algo(const torch::Tensor& x, const torch::Tensor& w,
              torch::Tensor& out, torch::Tensor& workspace,
              hipblasLtMatmulAlgo_t* algo, Layouts& L) {
  float alpha = 1.f, beta = 0.f;
  LT_CHECK(hipblasLtMatmul(
      lt_handle(), L.op, &alpha, w.data_ptr(), L.a, x.data_ptr(), L.b,
      &beta, out.data_ptr(), L.c, out.data_ptr(), L.c, algo,
      workspace.data_ptr(), (size_t)workspace.numel(),
      at::hip::getCurrentHIPStream()));
}

// Enumerate + time every supported solution. Returns (indices, ms) sorted
// fastest-first.
std::vector<torch::Tensor> gemm_lt_search(torch::Tensor x, torch::Tensor w,
                                          long iters, long cap) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.dim() == 2 && w.dim() == 2);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16);
  x = x.contiguous();
  w = w.contiguous();
  const long long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "w must be [N, K]");
  auto out = torch::empty({M, N}, x.options());
  auto workspace = torch::empty({128 << 20},
                                x.options().dtype(torch::kUInt8));

  Layouts L;
  make_layouts(L, M, N, K);

  std::vector<hipblasLtMatmulHeuristicResult_t> all;
  LT_CHECK(hipblaslt_ext::getAllAlgos(
      lt_handle(), hipblaslt_ext::GemmType::HIPBLASLT_GEMM,
      HIPBLAS_OP_T, HIPBLAS_OP_N, HIP_R_16BF, HIP_R_16BF, HIP_R_16BF,
      HIP_R_16BF, HIPBLAS_COMPUTE_32F, all));

  float alpha = 1.f, beta = 0.f;
  std::vector<int> idxs;
  std::vector<float> times;
  hipEvent_t ev0, ev1;
  HIPRT_CHECK(hipEventCreate(&ev0));
  HIPRT_CHECK(hipEventCreate(&ev1));
  auto stream = at::hip::getCurrentHIPStream();
  for (auto& h : all) {
    if (cap > 0 && (long)idxs.size() >= cap)
      break;
    size_t ws = 0;
    if (hipblaslt_ext::matmulIsAlgoSupported(
            lt_handle(), L.op, &alpha, L.a, L.b, &beta, L.c, L.c, h.algo,
            ws) != HIPBLAS_STATUS_SUCCESS)
      continue;
    if (ws > (size_t)workspace.numel())
      continue;
    // warmup x2, then time `iters` back-to-back launches
    run_algo(x, w, out, workspace, &h.algo, L);
    run_algo(x, w, out, workspace, &h.algo, L);
    HIPRT_CHECK(hipEventRecord(ev0, stream));
    for (long i = 0; i < iters; ++i)
      run_algo(x, w, out, workspace, &h.algo, L);
    HIPRT_CHECK(hipEventRecord(ev1, stream));
    HIPRT_CHECK(hipEventSynchronize(ev1));
    float ms = 0.f;
    HIPRT_CHECK(hipEventElapsedTime(&ms, ev0, ev1));
    idxs.push_back(hipblaslt_ext::getIndexFromAlgo(h.algo));
    times.push_back(ms / iters);
  }
  HIPRT_CHECK(hipEventDestroy(ev0));
  HIPRT_CHECK(hipEventDestroy(ev1));

  // sort fastest-first
  std::vector<size_t> order(idxs.size());
  for (size_t i = 0; i < order.size(); ++i) order[i] = i;
  std::sort(order.begin(), order.end(),
            [&](size_t a, size_t b) { return times[a] < times[b]; });
  auto t_idx = torch::empty({(long long)order.size()},
                            torch::dtype(torch::kInt64));
  auto t_ms = torch::empty({(long long)order.size()},
                           torch::dtype(torch::kFloat32));
  for (size_t i = 0; i < order.size(); ++i) {
    t_idx[i] = (long long)idxs[order[i]];
    t_ms[i] = times[order[i]];
  }
  return {t_idx, t_ms};
}

// Run one GEMM with a pinned solution index (from gemm_lt_search).
torch::Tensor gemm_lt_run(torch::Tensor x, torch::Tensor w, long index) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda());
  x = x.contiguous();
  w = w.contiguous();
  const long long M = x.size(0), K = x.size(1), N = w.size(0);
  auto out = torch::empty({M, N}, x.options());
  auto workspace = torch::empty({128 << 20},
                                x.options().dtype(torch::kUInt8));
  Layouts L;
  make_layouts(L, M, N, K);
  std::vector<int> want{(int)index};
  std::vector<hipblasLtMatmulHeuristicResult_t> algos;
  LT_CHECK(hipblaslt_ext::getAlgosFromIndex(lt_handle(), want, algos));
  TORCH_CHECK(!algos.empty(), "no algo for index ", index);
  float alpha = 1.f, beta = 0.f;
  size_t ws = 0;
  LT_CHECK(hipblaslt_ext::matmulIsAlgoSupported(
      lt_handle(), L.op, &alpha, L.a, L.b, &beta, L.c, L.c,
      algos[0].algo, ws));
  run_algo(x, w, out, workspace, &algos[0].algo, L);
  return out;
}

// hipBLASLt's own heuristic choice (what torch.matmul effectively uses),
// for A/B comparison in the search report.
torch::Tensor gemm_lt_heuristic(torch::Tensor x, torch::Tensor w) {
  x = x.contiguous();
  w = w.contiguous();
  const long long M = x.size(0), K = x.size(1), N = w.size(0);
  auto out = torch::empty({M, N}, x.options());
  auto workspace = torch::empty({128 << 20},
                                x.options().dtype(torch::kUInt8));
  Layouts L;
  make_layouts(L, M, N, K);
  hipblasLtMatmulPreference_t pref;
  LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  uint64_t wsmax = (uint64_t)workspace.numel();
  LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &wsmax,
      sizeof(wsmax)));
  hipblasLtMatmulHeuristicResult_t res[1];
  int found = 0;
  LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(lt_handle(), L.op, L.a, L.b,
                                           L.c, L.c, pref, 1, res,
                                           &found));
  hipblasLtMatmulPreferenceDestroy(pref);
  TORCH_CHECK(found > 0, "no heuristic solution");
  run_algo(x, w, out, workspace, &res[0].algo, L);
  return out;
}

// ---------------------------------------------------------------- fp8
// OCP E4M3 GEMM (gfx950 native; 2x bf16 peak): D_bf16 = (A_fp8 @ B_fp8^T)
// with fp32 accumulate. Inputs arrive as uint8 bit-patterns (torch's
// float8_e4m3fn storage viewed as uint8); scales fold into alpha.
// Round-2 measurement target — compile-validated in round 1.
void make_layouts_fp8(Layouts& L, long long M, long long N, long long K) {
  LT_CHECK(hipblasLtMatmulDescCreate(&L.op, HIPBLAS_COMPUTE_32F,
                                     HIP_R_32F));
  hipblasOperation_t t = HIPBLAS_OP_T, n = HIPBLAS_OP_N;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      L.op, HIPBLASLT_MATMUL_DESC_TRANSA, &t, sizeof(t)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      L.op, HIPBLASLT_MATMUL_DESC_TRANSB, &n, sizeof(n)));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&L.a, HIP_R_8F_E4M3, K, N, K));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&L.b, HIP_R_8F_E4M3, K, M, K));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&L.c, HIP_R_16BF, N, M, N));
}

torch::Tensor gemm_lt_fp8(torch::Tensor x8, torch::Tensor w8,
                          double alpha_scale) {
  TORCH_CHECK(x8.is_cuda() && w8.is_cuda());
  TORCH_CHECK(x8.scalar_type() == torch::kUInt8 &&
              w8.scalar_type() == torch::kUInt8,
              "pass float8_e4m3fn storage viewed as uint8");
  x8 = x8.contiguous();
  w8 = w8.contiguous();
  const long long M = x8.size(0), K = x8.size(1), N = w8.size(0);
  auto out = torch::empty({M, N},
                          x8.options().dtype(torch::kBFloat16));
  auto workspace = torch::empty({128 << 20},
                                x8.options().dtype(torch::kUInt8));
  Layouts L;
  make_layouts_fp8(L, M, N, K);
  hipblasLtMatmulPreference_t pref;
  LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  uint64_t wsmax = (uint64_t)workspace.numel();
  LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &wsmax,
      sizeof(wsmax)));
  hipblasLtMatmulHeuristicResult_t res[1];
  int found = 0;
  LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(lt_handle(), L.op, L.a, L.b,
                                           L.c, L.c, pref, 1, res,
                                           &found));
  hipblasLtMatmulPreferenceDestroy(pref);
  TORCH_CHECK(found > 0, "no fp8 solution for this shape");
  float alpha = (float)alpha_scale, beta = 0.f;
  LT_CHECK(hipblasLtMatmul(
      lt_handle(), L.op, &alpha, w8.data_ptr(), L.a, x8.data_ptr(), L.b,
      &beta, out.data_ptr(), L.c, out.data_ptr(), L.c, &res[0].algo,
      workspace.data_ptr(), (size_t)workspace.numel(),
      at::hip::getCurrentHIPStream()));
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("search", &gemm_lt_search,
        "time hipBLASLt solutions for out = x @ w^T (cap=0: all); "
        "returns (indices, ms) fastest-first");
  m.def("run", &gemm_lt_run, "GEMM with a pinned solution index");
  m.def("heuristic", &gemm_lt_heuristic, "GEMM via hipBLASLt heuristic");
  m.def("fp8", &gemm_lt_fp8,
        "OCP E4M3 GEMM -> bf16 out (uint8-viewed fp8 inputs, alpha "
        "carries the descale)");
}
