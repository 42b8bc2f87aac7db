// Common device helpers for the MI355X (gfx950 / CDNA4) kernels.
//
// Conventions (see /opt/skills/guides/cdna_hip_programming.md):
//   * wavefront = 64 lanes; block sizes are multiples of 64
//   * bf16 memory traffic is vectorized as ushort4/uint4 (8/16 B per lane)
//   * fp32 accumulation everywhere; bf16 only at the memory boundary
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64

using bf16_t = __hip_bfloat16;

// 8 bf16 = 16 bytes, the coalescing sweet spot for bf16 streams
typedef unsigned short ushort8_t __attribute__((ext_vector_type(8)));
typedef float floatx4 __attribute__((ext_vector_type(4)));

typedef short shortx8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  unsigned int x = (unsigned int)u << 16;
  return __uint_as_float(x);
}

__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  // hardware convert (v_cvt_pk_bf16_f32 when the compiler pairs them):
  // round-to-nearest-even, matching PyTorch's float->bfloat16 cast.
  // The manual bit-twiddled version costs ~7 VALU ops per element and
  // made the attention forward VALU-bound (MfmaUtil 3.5%, VALUBusy 47%).
  union {
    __hip_bfloat16 b;
    unsigned short u;
  } cv;
  cv.b = __float2bfloat16(f);
  return cv.u;
}

// full-wave sum: every lane ends with the 64-lane total
__device__ __forceinline__ float wave_all_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

__device__ __forceinline__ float wave_all_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// block-level reduction over NW waves (NW <= 16), using caller-provided
// LDS scratch of NW floats; every thread returns the block total
template <int NW>
__device__ __forceinline__ float block_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  v = wave_all_sum(v);
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < NW; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

template <int NW>
__device__ __forceinline__ float block_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  v = wave_all_max(v);
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  float total = -INFINITY;
#pragma unroll
  for (int i = 0; i < NW; ++i) total = fmaxf(total, scratch[i]);
  __syncthreads();
  return total;
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e_ = hipGetLastError();                                       \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel launch failed: ",              \
                hipGetErrorString(e_));                                      \
  } while (0)
