// hipBLASLt algorithm search for the serving GEMM shapes (round-2 lever:
// the r32 profile shows the library GEMMs at ~50-60 % MFU on the default
// heuristic picks — profiles/r01_bench512_kernelstats.md).
//
// Exposes the hipblaslt-ext enumeration so benchmarks/gemm_algo_search.py
// can time EVERY supported algorithm for a shape and record the winner's
// stable index (hipblaslt_ext::getIndexFromAlgo), which
// hipblaslt_gemm_indexed() can then replay.
//
// Math/layout: out_rm[M,N] = x_rm[M,K] @ w_rm[N,K]^T.  In hipBLASLt's
// column-major view: D_cm[N,M] = op_T(w_cm[K,N]) * op_N(x_cm[K,M]), all
// operands packed, fp32 accumulation on bf16 inputs (the serving dtype).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hipblaslt/hipblaslt-ext.hpp>

#include <tuple>
#include <vector>

#define HIPBLASLT_CHECK(x)                                                \
  do {                                                                    \
    hipblasStatus_t st_ = (x);                                            \
    TORCH_CHECK(st_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ",        \
                (int)st_, " at ", #x);                                    \
  } while (0)

namespace {

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    TORCH_CHECK(hipblasLtCreate(&hh) == HIPBLAS_STATUS_SUCCESS);
    return hh;
  }();
  return h;
}

hipblaslt_ext::Gemm make_gemm(torch::Tensor out, torch::Tensor x,
                              torch::Tensor w) {
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16 &&
              out.scalar_type() == torch::kBFloat16);
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N);
  hipblaslt_ext::Gemm gemm(lt_handle(), HIPBLAS_OP_T, HIPBLAS_OP_N,
                           HIP_R_16BF, HIP_R_16BF, HIP_R_16BF, HIP_R_16BF,
                           HIPBLAS_COMPUTE_32F);
  hipblaslt_ext::GemmEpilogue epilogue;  // default: no epilogue
  hipblaslt_ext::GemmInputs inputs;
  static float one = 1.f, zero = 0.f;
  inputs.setA(w.data_ptr());
  inputs.setB(x.data_ptr());
  inputs.setC(out.data_ptr());
  inputs.setD(out.data_ptr());
  inputs.setAlpha(&one);
  inputs.setBeta(&zero);
  HIPBLASLT_CHECK(gemm.setProblem(/*m=*/N, /*n=*/M, /*k=*/K,
                                  /*batch=*/1, epilogue, inputs));
  return gemm;
}

}  // namespace

std::vector<std::tuple<int64_t, double>> hipblaslt_algo_bench(
    torch::Tensor out, torch::Tensor x, torch::Tensor w, int64_t iters,
    int64_t max_algos) {
  // Times every supported heuristic algorithm; returns (algo_index, ms)
  // sorted as enumerated (caller sorts by ms).
  auto gemm = make_gemm(out, x, w);
  hipblaslt_ext::GemmPreference pref;
  constexpr size_t WS_CAP = 128u << 20;
  pref.setMaxWorkspaceBytes(WS_CAP);
  std::vector<hipblasLtMatmulHeuristicResult_t> results;
  hipblasStatus_t st =
      gemm.algoGetHeuristic((int)max_algos, pref, results);
  TORCH_CHECK(st == HIPBLAS_STATUS_SUCCESS, "algoGetHeuristic failed");
  auto ws = torch::empty({(int64_t)WS_CAP},
                         torch::dtype(torch::kUInt8).device(x.device()));
  hipStream_t stream = at::hip::getCurrentHIPStream().stream();
  hipEvent_t ev0, ev1;
  TORCH_CHECK(hipEventCreate(&ev0) == hipSuccess);
  TORCH_CHECK(hipEventCreate(&ev1) == hipSuccess);

  std::vector<std::tuple<int64_t, double>> timed;
  for (auto& r : results) {
    size_t need = 0;
    if (gemm.isAlgoSupported(r.algo, need) != HIPBLAS_STATUS_SUCCESS ||
        need > WS_CAP)
      continue;
    if (gemm.initialize(r.algo, ws.data_ptr()) != HIPBLAS_STATUS_SUCCESS)
      continue;
    for (int i = 0; i < 2; ++i)  // warmup
      if (gemm.run(stream) != HIPBLAS_STATUS_SUCCESS) goto next;
    TORCH_CHECK(hipEventRecord(ev0, stream) == hipSuccess);
    for (int64_t i = 0; i < iters; ++i) gemm.run(stream);
    TORCH_CHECK(hipEventRecord(ev1, stream) == hipSuccess);
    TORCH_CHECK(hipEventSynchronize(ev1) == hipSuccess);
    {
      float ms = 0.f;
      TORCH_CHECK(hipEventElapsedTime(&ms, ev0, ev1) == hipSuccess);
      timed.emplace_back(hipblaslt_ext::getIndexFromAlgo(r.algo),
                         (double)ms / iters);
    }
  next:;
  }
  (void)hipEventDestroy(ev0);
  (void)hipEventDestroy(ev1);
  return timed;
}

void hipblaslt_gemm_indexed(torch::Tensor out, torch::Tensor x,
                            torch::Tensor w, int64_t algo_index,
                            torch::Tensor workspace) {
  // Replay a previously searched algorithm by stable index.
  auto gemm = make_gemm(out, x, w);
  std::vector<int> idx{(int)algo_index};
  std::vector<hipblasLtMatmulHeuristicResult_t> results;
  HIPBLASLT_CHECK(
      hipblaslt_ext::getAlgosFromIndex(lt_handle(), idx, results));
  TORCH_CHECK(!results.empty(), "no algo at index ", algo_index);
  size_t need = 0;
  HIPBLASLT_CHECK(gemm.isAlgoSupported(results[0].algo, need));
  TORCH_CHECK((int64_t)need <= workspace.numel(), "workspace too small: ",
              need);
  HIPBLASLT_CHECK(gemm.initialize(results[0].algo, workspace.data_ptr()));
  HIPBLASLT_CHECK(gemm.run(at::hip::getCurrentHIPStream().stream()));
}
