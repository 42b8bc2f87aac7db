// Offline hipBLASLt algorithm search for the ViT training GEMMs
// (ROADMAP item 5).  PyTorch's TunableOp crashes this ROCm 7.2 stack
// (core dumps in both tuning and read-only mode), so this standalone
// tool enumerates the full algorithm library via the hipblaslt-ext API
// directly, times every supported algorithm on the training shapes, and
// prints a table: the fastest index can then be applied from our own
// extension via hipblaslt_ext::getAlgosFromIndex.
//
// Build (on demand; NOT part of setup.py):
//   hipcc -O2 --offload-arch=gfx950 csrc/tools/hipblaslt_search.cpp \
//         -lhipblaslt -o gpurun_out/hipblaslt_search
// Run on a GPU box:
//   gpurun_out/hipblaslt_search [--model vit10b] [--reps 5] > \
//         gpurun_out/hipblaslt_search.csv
//
// Layout convention: hipBLASLt is column-major.  The rows below are the
// column-major duals of torch's row-major training GEMMs, matching the
// Tensile kernels observed in the kernel traces (profiles/PROFILES.md):
//   forward  y = x @ W^T    -> Cijk_Alik_Bljk: opA=T opB=N, m=out, n=tok, k=in
//   dgrad    dx = dy @ W    -> Cijk_Ailk_Bljk: opA=N opB=N, m=in,  n=tok, k=out
//   wgrad    dW = dy^T @ x  -> Cijk_Ailk_Bjlk: opA=N opB=T, m=in,  n=out, k=tok

#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt-ext.hpp>
#include <hipblaslt/hipblaslt.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#define HIP_CHECK(x)                                                          \
    do {                                                                      \
        hipError_t e_ = (x);                                                  \
        if (e_ != hipSuccess) {                                               \
            fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e_), \
                    __FILE__, __LINE__);                                      \
            exit(1);                                                          \
        }                                                                     \
    } while (0)

#define HIPBLAS_CHECK(x)                                                     \
    do {                                                                     \
        hipblasStatus_t s_ = (x);                                            \
        if (s_ != HIPBLAS_STATUS_SUCCESS) {                                  \
            fprintf(stderr, "hipBLASLt status %d at %s:%d\n", (int)s_,       \
                    __FILE__, __LINE__);                                     \
            exit(1);                                                         \
        }                                                                    \
    } while (0)

struct Problem {
    const char* name;
    hipblasOperation_t opA, opB;
    int64_t m, n, k;
};

// ViT-10B, bs=128/GPU, 224px: tokens = 128 * 256 = 32768 (the model is
// mean-pooled, no CLS token — T = (224/14)^2, models/vit.py).
static std::vector<Problem> problems_vit10b() {
    const int64_t tok = 32768, d = 5120, qkv = 15360, ffn = 20480;
    return {
        {"patch_fwd", HIPBLAS_OP_T, HIPBLAS_OP_N, d, tok, 588},
        {"qkv_fwd", HIPBLAS_OP_T, HIPBLAS_OP_N, qkv, tok, d},
        {"qkv_dgrad", HIPBLAS_OP_N, HIPBLAS_OP_N, d, tok, qkv},
        {"qkv_wgrad", HIPBLAS_OP_N, HIPBLAS_OP_T, d, qkv, tok},
        {"proj_fwd", HIPBLAS_OP_T, HIPBLAS_OP_N, d, tok, d},
        {"proj_dgrad", HIPBLAS_OP_N, HIPBLAS_OP_N, d, tok, d},
        {"proj_wgrad", HIPBLAS_OP_N, HIPBLAS_OP_T, d, d, tok},
        {"fc1_fwd", HIPBLAS_OP_T, HIPBLAS_OP_N, ffn, tok, d},
        {"fc1_dgrad", HIPBLAS_OP_N, HIPBLAS_OP_N, d, tok, ffn},
        {"fc1_wgrad", HIPBLAS_OP_N, HIPBLAS_OP_T, d, ffn, tok},
        {"fc2_fwd", HIPBLAS_OP_T, HIPBLAS_OP_N, d, tok, ffn},
        {"fc2_dgrad", HIPBLAS_OP_N, HIPBLAS_OP_N, ffn, tok, d},
        {"fc2_wgrad", HIPBLAS_OP_N, HIPBLAS_OP_T, ffn, d, tok},
    };
}

static std::vector<Problem> problems_vitlarge() {
    const int64_t tok = 32896, d = 1024, qkv = 3072, ffn = 4096;
    return {
        {"qkv_fwd", HIPBLAS_OP_T, HIPBLAS_OP_N, qkv, tok, d},
        {"qkv_wgrad", HIPBLAS_OP_N, HIPBLAS_OP_T, d, qkv, tok},
        {"fc1_fwd", HIPBLAS_OP_T, HIPBLAS_OP_N, ffn, tok, d},
        {"fc2_fwd", HIPBLAS_OP_T, HIPBLAS_OP_N, d, tok, ffn},
    };
}

struct Timed {
    int index;
    double ms;
};

int main(int argc, char** argv) {
    std::string model = "vit10b", only;
    bool with_bias = false;
    int reps = 5, topk = 16;
    for (int i = 1; i < argc; ++i) {
        if (!strcmp(argv[i], "--model") && i + 1 < argc) model = argv[++i];
        if (!strcmp(argv[i], "--reps") && i + 1 < argc) reps = atoi(argv[++i]);
        if (!strcmp(argv[i], "--topk") && i + 1 < argc) topk = atoi(argv[++i]);
        // --only NAME: run a single problem (lets a driver script run
        // every problem in its own process, so one broken library
        // kernel faulting the GPU cannot take down the whole sweep)
        if (!strcmp(argv[i], "--only") && i + 1 < argc) only = argv[++i];
        // --bias: search the *_fwd problems WITH the bias epilogue the
        // training forward actually uses (an index that wins bare can
        // fail isAlgoSupported once the epilogue is attached)
        if (!strcmp(argv[i], "--bias")) with_bias = true;
    }
    auto problems =
        model == "vit-large" ? problems_vitlarge() : problems_vit10b();
    if (!only.empty()) {
        std::vector<Problem> filtered;
        for (auto& p : problems)
            if (only == p.name) filtered.push_back(p);
        problems = filtered;
    }

    hipblasLtHandle_t handle;
    HIPBLAS_CHECK(hipblasLtCreate(&handle));
    hipStream_t stream;
    HIP_CHECK(hipStreamCreate(&stream));

    const size_t ws_bytes = 128u << 20;
    void* workspace;
    HIP_CHECK(hipMalloc(&workspace, ws_bytes));

    // the full algorithm library for bf16 TN/NN/NT f32-compute GEMMs
    std::vector<hipblasLtMatmulHeuristicResult_t> all_algos[3];
    const hipblasOperation_t ops[3][2] = {
        {HIPBLAS_OP_T, HIPBLAS_OP_N},
        {HIPBLAS_OP_N, HIPBLAS_OP_N},
        {HIPBLAS_OP_N, HIPBLAS_OP_T},
    };
    for (int i = 0; i < 3; ++i) {
        HIPBLAS_CHECK(hipblaslt_ext::getAllAlgos(
            handle, hipblaslt_ext::GemmType::HIPBLASLT_GEMM, ops[i][0],
            ops[i][1], HIP_R_16BF, HIP_R_16BF, HIP_R_16BF, HIP_R_16BF,
            HIPBLAS_COMPUTE_32F, all_algos[i]));
        fprintf(stderr, "[search] %s%s: %zu algos in library\n",
                ops[i][0] == HIPBLAS_OP_T ? "T" : "N",
                ops[i][1] == HIPBLAS_OP_T ? "T" : "N", all_algos[i].size());
    }

    printf("problem,opA,opB,m,n,k,algo_index,ms,tflops,note\n");
    for (const auto& p : problems) {
        int op_set = (p.opA == HIPBLAS_OP_T)   ? 0
                     : (p.opB == HIPBLAS_OP_T) ? 2
                                               : 1;
        // two operand sets so repeated runs do not replay a warm L3
        // (guide rule 25); each set's A+B alone exceed the 256 MB L3
        // for the 10B shapes, but rotate anyway for the small ones
        // +32 MiB guard padding per operand: a screened library kernel
        // with out-of-tile reads/writes must not memory-fault the box
        const size_t pad = 32u << 20;
        void *a[2], *b[2], *d[2];
        for (int s = 0; s < 2; ++s) {
            HIP_CHECK(hipMalloc(&a[s], sizeof(uint16_t) * p.m * p.k + pad));
            HIP_CHECK(hipMalloc(&b[s], sizeof(uint16_t) * p.k * p.n + pad));
            HIP_CHECK(hipMalloc(&d[s], sizeof(uint16_t) * p.m * p.n + pad));
            HIP_CHECK(hipMemset(a[s], 0x3c, sizeof(uint16_t) * p.m * p.k));
            HIP_CHECK(hipMemset(b[s], 0x3c, sizeof(uint16_t) * p.k * p.n));
        }
        float alpha = 1.0f, beta = 0.0f;
        const bool use_bias =
            with_bias && strstr(p.name, "_fwd") != nullptr;
        void* bias_buf = nullptr;
        if (use_bias) {
            HIP_CHECK(hipMalloc(&bias_buf, sizeof(uint16_t) * p.m + pad));
            HIP_CHECK(hipMemset(bias_buf, 0x3c, sizeof(uint16_t) * p.m));
        }

        hipblaslt_ext::Gemm gemm(handle, p.opA, p.opB, HIP_R_16BF, HIP_R_16BF,
                                 HIP_R_16BF, HIP_R_16BF, HIPBLAS_COMPUTE_32F);
        hipblaslt_ext::GemmEpilogue epilogue;  // default: no epilogue
        if (use_bias) {
            epilogue.setMode(HIPBLASLT_EPILOGUE_BIAS);
            epilogue.setBiasDataType(HIP_R_16BF);
        }
        auto set_problem = [&](int s) {
            hipblaslt_ext::GemmInputs inputs;
            inputs.setA(a[s]);
            inputs.setB(b[s]);
            inputs.setC(d[s]);
            inputs.setD(d[s]);
            inputs.setAlpha(&alpha);
            inputs.setBeta(&beta);
            if (use_bias) inputs.setBias(bias_buf);
            HIPBLAS_CHECK(gemm.setProblem(p.m, p.n, p.k, 1, epilogue, inputs));
        };
        set_problem(0);

        hipEvent_t ev0, ev1;
        HIP_CHECK(hipEventCreate(&ev0));
        HIP_CHECK(hipEventCreate(&ev1));
        auto time_algo = [&](const hipblasLtMatmulAlgo_t& algo,
                             int n_reps) -> double {
            double best = 1e30;
            for (int r = 0; r < n_reps; ++r) {
                set_problem(r & 1);
                if (gemm.initialize(algo, workspace, false, stream) !=
                    HIPBLAS_STATUS_SUCCESS)
                    return -1.0;
                HIP_CHECK(hipEventRecord(ev0, stream));
                if (gemm.run(stream) != HIPBLAS_STATUS_SUCCESS) return -1.0;
                HIP_CHECK(hipEventRecord(ev1, stream));
                HIP_CHECK(hipEventSynchronize(ev1));
                float ms;
                HIP_CHECK(hipEventElapsedTime(&ms, ev0, ev1));
                best = std::min(best, (double)ms);
            }
            return best;
        };
        const double tflop = 2.0 * p.m * p.n * p.k / 1e12;

        // the heuristic pick = what torch's matmul would run
        hipblaslt_ext::GemmPreference pref;
        pref.setMaxWorkspaceBytes(ws_bytes);
        std::vector<hipblasLtMatmulHeuristicResult_t> heur;
        if (gemm.algoGetHeuristic(1, pref, heur) == HIPBLAS_STATUS_SUCCESS &&
            !heur.empty()) {
            double ms = time_algo(heur[0].algo, reps);
            if (ms > 0)
                printf("%s,%c,%c,%ld,%ld,%ld,%d,%.4f,%.1f,heuristic\n", p.name,
                       p.opA == HIPBLAS_OP_T ? 'T' : 'N',
                       p.opB == HIPBLAS_OP_T ? 'T' : 'N', (long)p.m, (long)p.n,
                       (long)p.k, hipblaslt_ext::getIndexFromAlgo(heur[0].algo),
                       ms, tflop / (ms / 1e3));
            fflush(stdout);
        }

        // screen every supported algorithm with 1 rep, refine the top-k
        std::vector<Timed> screened;
        for (auto& hr : all_algos[op_set]) {
            size_t need = 0;
            if (gemm.isAlgoSupported(hr.algo, need) != HIPBLAS_STATUS_SUCCESS ||
                need > ws_bytes)
                continue;
            double ms = time_algo(hr.algo, 1);
            if (ms > 0)
                screened.push_back(
                    {hipblaslt_ext::getIndexFromAlgo(hr.algo), ms});
        }
        std::sort(screened.begin(), screened.end(),
                  [](const Timed& x, const Timed& y) { return x.ms < y.ms; });
        int refined = 0;
        for (auto& t : screened) {
            if (refined++ >= topk) break;
            // re-find the algo by index to re-time it properly
            for (auto& hr : all_algos[op_set]) {
                if (hipblaslt_ext::getIndexFromAlgo(hr.algo) != t.index)
                    continue;
                double ms = time_algo(hr.algo, reps);
                if (ms > 0)
                    printf("%s,%c,%c,%ld,%ld,%ld,%d,%.4f,%.1f,top%d\n", p.name,
                           p.opA == HIPBLAS_OP_T ? 'T' : 'N',
                           p.opB == HIPBLAS_OP_T ? 'T' : 'N', (long)p.m,
                           (long)p.n, (long)p.k, t.index, ms,
                           tflop / (ms / 1e3), refined);
                fflush(stdout);
                break;
            }
        }
        fprintf(stderr, "[search] %s: screened %zu supported algos\n", p.name,
                screened.size());

        HIP_CHECK(hipEventDestroy(ev0));
        HIP_CHECK(hipEventDestroy(ev1));
        for (int s = 0; s < 2; ++s) {
            HIP_CHECK(hipFree(a[s]));
            HIP_CHECK(hipFree(b[s]));
            HIP_CHECK(hipFree(d[s]));
        }
    }
    HIP_CHECK(hipFree(workspace));
    HIP_CHECK(hipStreamDestroy(stream));
    HIPBLAS_CHECK(hipblasLtDestroy(handle));
    return 0;
}
