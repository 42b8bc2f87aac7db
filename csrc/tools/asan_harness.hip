// Device-AddressSanitizer RUN harness (SURVEY.md §5 sanitizers;
// VERDICT r1 weak #6): launches the raw-pointer HIP kernels standalone
// (no torch) with instrumentation active, including non-multiple tile
// shapes that stress every bounds check.
//
// One kernel family per binary (the families' anonymous namespaces
// collide if included together):
//   hipcc --offload-arch=gfx950:xnack+ -fsanitize=address -g -O1 \
//     -std=c++17 -DHARNESS_FMHA csrc/tools/asan_harness.hip -o /tmp/h_fmha
//   ... -DHARNESS_WGEMM ... / ... -DHARNESS_FGEMM ...
// Run on a GPU box with HSA_XNACK=1.  Exit 0 = no ASAN report.

#define VITFSDP_KERNELS_ONLY 1

#include <cstdio>
#include <cstdlib>
#include <vector>

#if defined(HARNESS_FMHA)
#include "../fmha.hip"
#elif defined(HARNESS_WGEMM)
#include "../wgemm.hip"
#elif defined(HARNESS_FGEMM)
#include "../fgemm.hip"
#else
#error "define one of HARNESS_FMHA / HARNESS_WGEMM / HARNESS_FGEMM"
#endif

#define CHECK(x)                                                      \
  do {                                                                \
    hipError_t e_ = (x);                                              \
    if (e_ != hipSuccess) {                                           \
      fprintf(stderr, "HIP error %s at %s:%d\n",                      \
              hipGetErrorString(e_), __FILE__, __LINE__);             \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

template <typename T>
T* alloc_fill(size_t n, unsigned short pat = 0x3c00) {
  T* p;
  CHECK(hipMalloc(&p, n * sizeof(T)));
  CHECK(hipMemsetD16((hipDeviceptr_t)p, pat, n * sizeof(T) / 2));
  return p;
}

int main() {
#if defined(HARNESS_FMHA)
  // T=200 is deliberately NOT a tile multiple: exercises every q/k
  // bounds guard under ASAN
  const int B = 2, H = 2, T = 200, D = 160;
  using S = FmhaShapes<160>;
  QkvStrides st;
  st.qt = D;
  st.qh = (long)T * D;
  st.qb = (long)H * st.qh;
  st.ot = D;
  st.oh = (long)T * D;
  st.ob = (long)H * st.oh;
  const size_t n = (size_t)B * H * T * D;
  short* q = alloc_fill<short>(n);
  short* k = alloc_fill<short>(n);
  short* v = alloc_fill<short>(n);
  short* o = alloc_fill<short>(n);
  short* dq = alloc_fill<short>(n);
  short* dk = alloc_fill<short>(n);
  short* dv = alloc_fill<short>(n);
  float *lse, *delta;
  CHECK(hipMalloc(&lse, sizeof(float) * B * H * T));
  CHECK(hipMemset(lse, 0, sizeof(float) * B * H * T));
  CHECK(hipMalloc(&delta, sizeof(float) * B * H * T));
  CHECK(hipMemset(delta, 0, sizeof(float) * B * H * T));

  CHECK(hipFuncSetAttribute(
      reinterpret_cast<const void*>(&fmha_fwd_kernel<160>),
      hipFuncAttributeMaxDynamicSharedMemorySize, sizeof(S::Shared)));
  CHECK(hipFuncSetAttribute(
      reinterpret_cast<const void*>(&fmha_bwd_dq_kernel<160>),
      hipFuncAttributeMaxDynamicSharedMemorySize, sizeof(S::SharedDQ)));
  CHECK(hipFuncSetAttribute(
      reinterpret_cast<const void*>(&fmha_bwd_dkv_kernel<160>),
      hipFuncAttributeMaxDynamicSharedMemorySize, sizeof(S::SharedDKV)));

  dim3 gf((unsigned)(B * H), (T + S::QTile - 1) / S::QTile);
  hipLaunchKernelGGL(fmha_fwd_kernel<160>, gf, dim3(kBlockThreads),
                     sizeof(S::Shared), 0, q, k, v, o, lse, T, H, st, 0.1f);
  CHECK(hipGetLastError());
  CHECK(hipDeviceSynchronize());
  printf("fmha_fwd ok\n");

  hipLaunchKernelGGL(fmha_bwd_dq_kernel<160>, gf, dim3(kBlockThreads),
                     sizeof(S::SharedDQ), 0, q, k, v, o /*dO*/, lse, delta,
                     dq, T, H, st, 0.1f);
  CHECK(hipGetLastError());
  CHECK(hipDeviceSynchronize());
  printf("fmha_bwd_dq ok\n");

  dim3 gkv((unsigned)(B * H), (T + 63) / 64);
  hipLaunchKernelGGL(fmha_bwd_dkv_kernel<160>, gkv, dim3(kBlockThreads),
                     sizeof(S::SharedDKV), 0, q, k, v, o /*dO*/, lse, delta,
                     dk, dv, T, H, st, 0.1f);
  CHECK(hipGetLastError());
  CHECK(hipDeviceSynchronize());
  printf("fmha_bwd_dkv ok\n");

#elif defined(HARNESS_WGEMM)
  const int K = 512, M = 256, N = 512;
  short* a = alloc_fill<short>((size_t)K * M);
  short* b = alloc_fill<short>((size_t)K * N);
  short* c = alloc_fill<short>((size_t)M * N);
  float* dbias;
  CHECK(hipMalloc(&dbias, sizeof(float) * M));
  CHECK(hipMemset(dbias, 0, sizeof(float) * M));
  CHECK(hipFuncSetAttribute(
      reinterpret_cast<const void*>(&wgemm_atb_kernel<false>),
      hipFuncAttributeMaxDynamicSharedMemorySize, sizeof(WgemmShared)));
  dim3 grid((unsigned)(M / 256), (unsigned)(N / 256));
  hipLaunchKernelGGL(wgemm_atb_kernel<false>, grid, dim3(512),
                     sizeof(WgemmShared), 0, a, b, c, dbias, K, M, N);
  CHECK(hipGetLastError());
  CHECK(hipDeviceSynchronize());
  printf("wgemm ok\n");

#elif defined(HARNESS_FGEMM)
  const int M = 512, N = 256, K = 128;
  short* x = alloc_fill<short>((size_t)M * K);
  short* w = alloc_fill<short>((size_t)N * K);
  short* c = alloc_fill<short>((size_t)M * N);
  short* aux = alloc_fill<short>((size_t)M * N);
  short* bias = alloc_fill<short>((size_t)N);
  for (auto fn : {reinterpret_cast<const void*>(&fgemm_abt_kernel<0, 64>),
                  reinterpret_cast<const void*>(&fgemm_abt_kernel<2, 64>)}) {
    CHECK(hipFuncSetAttribute(fn, hipFuncAttributeMaxDynamicSharedMemorySize,
                              sizeof(FgemmShared<64>)));
  }
  dim3 grid((unsigned)(M / 256), (unsigned)(N / 256));
  hipLaunchKernelGGL((fgemm_abt_kernel<0, 64>), grid, dim3(512),
                     sizeof(FgemmShared<64>), 0, x, w, nullptr, c, nullptr, M, N,
                     K);
  CHECK(hipGetLastError());
  CHECK(hipDeviceSynchronize());
  hipLaunchKernelGGL((fgemm_abt_kernel<2, 64>), grid, dim3(512),
                     sizeof(FgemmShared<64>), 0, x, w, bias, c, aux, M, N, K);
  CHECK(hipGetLastError());
  CHECK(hipDeviceSynchronize());
  printf("fgemm ok\n");
#endif
  printf("ASAN HARNESS PASS\n");
  return 0;
}
