// hipBLASLt invocation with an explicit algorithm index (ROADMAP item 5).
//
// PyTorch's TunableOp crashes this ROCm 7.2 stack, so algorithm
// selection is done offline by csrc/tools/hipblaslt_search.cpp; this
// host-side entry applies a chosen index from Python:
//
//   d = _C.lt_gemm(a, b, algo_index)   # row-major a[M,K] @ b[K,N]
//
// algo_index < 0 uses the library heuristic (same pick as torch's
// matmul) — giving a GPU test that validates the whole hipblaslt-ext
// path (and the row/column-major duality) against torch.matmul before
// any tuned index is wired in.
//
// Row-major C = A @ B is computed as the column-major dual
// C^T = B^T A^T: opA = opB = N, (m, n, k) = (N, M, K), A-ptr = b,
// B-ptr = a, leading dims (N, K, N).

#ifndef VITFSDP_KERNELS_ONLY
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include <hipblaslt/hipblaslt-ext.hpp>
#include <hipblaslt/hipblaslt.h>

#include <mutex>
#include <stdexcept>
#include <vector>

namespace {

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t handle = [] {
    hipblasLtHandle_t h;
    if (hipblasLtCreate(&h) != HIPBLAS_STATUS_SUCCESS)
      throw std::runtime_error("hipblasLtCreate failed");
    return h;
  }();
  return handle;
}

void* lt_workspace(size_t bytes) {
  static void* ws = nullptr;
  static size_t ws_bytes = 0;
  static std::mutex mu;
  std::lock_guard<std::mutex> lock(mu);
  if (bytes > ws_bytes) {
    if (ws) (void)hipFree(ws);
    if (hipMalloc(&ws, bytes) != hipSuccess)
      throw std::runtime_error("lt_gemm workspace alloc failed");
    ws_bytes = bytes;
  }
  return ws;
}

constexpr size_t kMaxWorkspace = 128u << 20;

}  // namespace

torch::Tensor lt_gemm(torch::Tensor a, torch::Tensor b, long algo_index,
                      c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "lt_gemm: CUDA tensors required");
  TORCH_CHECK(a.dtype() == torch::kBFloat16 && b.dtype() == torch::kBFloat16,
              "lt_gemm: bf16 only");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(0),
              "lt_gemm: a[M,K] @ b[K,N] expected");
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  const int64_t M = ac.size(0), K = ac.size(1), N = bc.size(1);
  auto d = torch::empty({M, N}, ac.options());

  auto handle = lt_handle();
  hipStream_t stream = at::cuda::getCurrentCUDAStream();

  hipblaslt_ext::Gemm gemm(handle, HIPBLAS_OP_N, HIPBLAS_OP_N, HIP_R_16BF,
                           HIP_R_16BF, HIP_R_16BF, HIP_R_16BF,
                           HIPBLAS_COMPUTE_32F);
  float alpha = 1.0f, beta = 0.0f;
  hipblaslt_ext::GemmEpilogue epilogue;
  hipblaslt_ext::GemmInputs inputs;
  torch::Tensor bias_c;
  if (bias.has_value()) {
    // per-output-feature bias: length N = rows of the column-major dual,
    // which is exactly hipBLASLt's bias-vector broadcast
    bias_c = bias->contiguous();
    TORCH_CHECK(bias_c.is_cuda() && bias_c.dtype() == torch::kBFloat16 &&
                    bias_c.numel() == N,
                "lt_gemm: bias must be bf16 CUDA of length N");
    epilogue.setMode(HIPBLASLT_EPILOGUE_BIAS);
    epilogue.setBiasDataType(HIP_R_16BF);
    inputs.setBias(bias_c.data_ptr());
  }
  inputs.setA(bc.data_ptr());  // column-major dual: A <- b
  inputs.setB(ac.data_ptr());
  inputs.setC(d.data_ptr());
  inputs.setD(d.data_ptr());
  inputs.setAlpha(&alpha);
  inputs.setBeta(&beta);
  // (m, n, k) = (N, M, K); short setProblem overload derives the
  // contiguous leading dims (N, K, N)
  TORCH_CHECK(
      gemm.setProblem(N, M, K, 1, epilogue, inputs) == HIPBLAS_STATUS_SUCCESS,
      "lt_gemm: setProblem failed");

  hipblasLtMatmulAlgo_t algo;
  bool have_algo = false;
  if (algo_index >= 0) {
    std::vector<hipblasLtMatmulHeuristicResult_t> found;
    std::vector<int> idx{static_cast<int>(algo_index)};
    if (hipblaslt_ext::getAlgosFromIndex(handle, idx, found) ==
            HIPBLAS_STATUS_SUCCESS &&
        !found.empty()) {
      size_t need = 0;
      if (gemm.isAlgoSupported(found[0].algo, need) ==
              HIPBLAS_STATUS_SUCCESS &&
          need <= kMaxWorkspace) {
        algo = found[0].algo;
        have_algo = true;
      }
    }
    TORCH_CHECK(have_algo, "lt_gemm: algo index ", algo_index,
                " not valid for this problem");
  } else {
    hipblaslt_ext::GemmPreference pref;
    pref.setMaxWorkspaceBytes(kMaxWorkspace);
    std::vector<hipblasLtMatmulHeuristicResult_t> heur;
    TORCH_CHECK(gemm.algoGetHeuristic(1, pref, heur) ==
                        HIPBLAS_STATUS_SUCCESS &&
                    !heur.empty(),
                "lt_gemm: no heuristic algorithm found");
    algo = heur[0].algo;
    have_algo = true;
  }

  TORCH_CHECK(gemm.initialize(algo, lt_workspace(kMaxWorkspace), false,
                              stream) == HIPBLAS_STATUS_SUCCESS,
              "lt_gemm: initialize failed");
  TORCH_CHECK(gemm.run(stream) == HIPBLAS_STATUS_SUCCESS,
              "lt_gemm: run failed");
  return d;
}

#endif  // VITFSDP_KERNELS_ONLY
