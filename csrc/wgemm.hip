// Weight-gradient GEMM for CDNA4: C[M,N] = A^T B with A = dY [K, M] and
// B = X [K, N] both row-major bf16 (the Linear-backward wgrad shape,
// K = batch*tokens = 32768 for the ViT-10B config).
//
// hipBLASLt runs these transposed-A shapes at ~1.0-1.1 PF/s vs ~1.5 for
// the forward TN GEMMs.  The hard part of a hand-written version is that
// the contraction dim K is the non-contiguous axis of both operands; the
// usual fix (transpose during LDS staging) costs 64 scalar LDS writes
// per thread per K-step.  gfx950's ds_read_b64_tr_b16 removes that
// entirely: tiles stay row-major in LDS (vectorized 16-B stores straight
// from the coalesced global loads) and the MFMA fragments are gathered
// along K by the hardware transpose read.
//
// ds_read_b64_tr_b16 semantics (pinned on-device by tr16_probe,
// tests/test_gpu_kernels.py): per 16-lane group, lane i reads 4
// contiguous u16 at its own 8-byte-aligned address; the 64 gathered
// elements form a stream ordered (lane, elem) and lane l receives
// stream[(l&15) + 16*j], j = 0..3.  Reading a [4 row][16 col] sub-tile
// with lane addresses row = base+(i>>2), col = colbase+(i&3)*4 therefore
// delivers column (l&15) of the sub-tile to lane l — a free transpose.
//
// Tiling: 256x256 C tile per 512-thread block (8 waves as 2m x 4n, each
// wave 128x64), BK = 64, single-buffered LDS (66.6 KB -> 2 blocks/CU)
// with T14 register staging: the next K-step's global loads are issued
// before this step's MFMA phases.

#ifndef VITFSDP_KERNELS_ONLY
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>
#endif

#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2;

constexpr int kBM = 256;
constexpr int kBN = 256;
constexpr int kBK = 64;
constexpr int kThreads = 512;  // 8 waves: 2 (m) x 4 (n)
// row stride of the LDS tiles in elements; +4 keeps every tr-read lane
// address 8-byte aligned while breaking the 256-element power-of-two
constexpr int kSA = kBM + 4;
constexpr int kSB = kBN + 4;

struct WgemmShared {
  short a_tile[kBK][kSA];  // dY tile, row-major [k][m]
  short b_tile[kBK][kSB];  // X  tile, row-major [k][n]
};

// one hardware transpose read: returns 4 u16 along rows base..base+3 of
// a row-major LDS tile at column (colbase + (lane&15))
__device__ __forceinline__ u32x2 tr_read(unsigned lds_byte_base, int lane15,
                                         int row_base, int col_base,
                                         int row_stride) {
  const unsigned addr =
      lds_byte_base +
      2u * ((unsigned)((row_base + (lane15 >> 2)) * row_stride + col_base +
                       (lane15 & 3) * 4));
  u32x2 v;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0" : "=v"(v) : "v"(addr));
  return v;
}

__global__ __launch_bounds__(kThreads, 2) void wgemm_atb_kernel(
    const short* __restrict__ a,  // [K, M]
    const short* __restrict__ b,  // [K, N]
    short* __restrict__ c,        // [M, N]
    float* __restrict__ dbias,    // optional [M] fp32 (atomic), may be null
    int K, int M, int N) {
  HIP_DYNAMIC_SHARED(char, smem_raw)
  WgemmShared& sm = *reinterpret_cast<WgemmShared*>(smem_raw);
  const unsigned a_base = (unsigned)__builtin_amdgcn_groupstaticsize();
  const unsigned b_base = a_base + sizeof(sm.a_tile);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int col = lane & 15;
  const int seg = lane >> 4;
  const int wm = wave >> 2;  // 0..1: wave's 128-row m strip
  const int wn = wave & 3;   // 0..3: wave's 64-col n strip

  const long m0 = (long)blockIdx.x * kBM;
  const long n0 = (long)blockIdx.y * kBN;

  // staging coordinates: 16-B vectors of the [64][256] tiles
  constexpr int kVecs = kBK * (kBM / 8) / kThreads;  // 4
  int s_kr[kVecs], s_c8[kVecs];
#pragma unroll
  for (int i = 0; i < kVecs; ++i) {
    const int idx = tid + i * kThreads;
    s_kr[i] = idx / (kBM / 8);
    s_c8[i] = (idx % (kBM / 8)) * 8;
  }
  bf16x8 a_st[kVecs], b_st[kVecs];
  auto issue_loads = [&](long k_base) {
#pragma unroll
    for (int i = 0; i < kVecs; ++i) {
      const long krow = k_base + s_kr[i];
      a_st[i] = *reinterpret_cast<const bf16x8*>(&a[krow * M + m0 + s_c8[i]]);
      b_st[i] = *reinterpret_cast<const bf16x8*>(&b[krow * N + n0 + s_c8[i]]);
    }
  };
  auto write_tiles = [&]() {
#pragma unroll
    for (int i = 0; i < kVecs; ++i) {
      *reinterpret_cast<bf16x8*>(&sm.a_tile[s_kr[i]][s_c8[i]]) = a_st[i];
      *reinterpret_cast<bf16x8*>(&sm.b_tile[s_kr[i]][s_c8[i]]) = b_st[i];
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  float bias_acc[8];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) bias_acc[mi] = 0.f;

  const int n_ksteps = K / kBK;
  issue_loads(0);
  write_tiles();
  __syncthreads();

  for (int ks = 0; ks < n_ksteps; ++ks) {
    if (ks + 1 < n_ksteps) issue_loads((long)(ks + 1) * kBK);

#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      // gather this chunk's fragments via hardware transpose reads:
      // A fragment mi: lane -> A_lds[kc*32+seg*8+j][wm*128+mi*16+col]
      u32x2 af[8][2], bf[4][2];
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
        for (int h = 0; h < 2; ++h) {
          af[mi][h] = tr_read(a_base, col, kc * 32 + seg * 8 + 4 * h,
                              wm * 128 + mi * 16, kSA);
        }
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
#pragma unroll
        for (int h = 0; h < 2; ++h) {
          bf[ni][h] = tr_read(b_base, col, kc * 32 + seg * 8 + 4 * h,
                              wn * 64 + ni * 16, kSB);
        }
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);

#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
        union {
          u32x2 u[2];
          bf16x8 v;
        } ac;
        ac.u[0] = af[mi][0];
        ac.u[1] = af[mi][1];
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          union {
            u32x2 u[2];
            bf16x8 v;
          } bc;
          bc.u[0] = bf[ni][0];
          bc.u[1] = bf[ni][1];
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              ac.v, bc.v, acc[mi][ni], 0, 0, 0);
        }
        if (dbias && wn == 0) {
          // dbias[m] = sum_k dY[k][m]: this lane's A fragment holds 8
          // consecutive k for column m = wm*128 + mi*16 + col
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            bias_acc[mi] += bf16_to_f32((unsigned short)(af[mi][h][0] & 0xffff)) +
                            bf16_to_f32((unsigned short)(af[mi][h][0] >> 16)) +
                            bf16_to_f32((unsigned short)(af[mi][h][1] & 0xffff)) +
                            bf16_to_f32((unsigned short)(af[mi][h][1] >> 16));
          }
        }
      }
    }

    __syncthreads();  // everyone is done reading this K-step's tiles
    if (ks + 1 < n_ksteps) {
      write_tiles();
      __syncthreads();  // next tiles staged
    }
  }

  // epilogue: C[m][n], C-layout rows m = seg*4+r
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long m = m0 + wm * 128 + mi * 16 + seg * 4 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const long n = n0 + wn * 64 + ni * 16 + col;
        c[m * N + n] = (short)f32_to_bf16(acc[mi][ni][r]);
      }
    }
    if (dbias && wn == 0) {
      // one lane per (m-column, seg-k-range): sum the 4 seg partials
      float v = bias_acc[mi];
      v += __shfl_xor(v, 16);
      v += __shfl_xor(v, 32);
      if (seg == 0) {
        atomicAdd(&dbias[m0 + wm * 128 + mi * 16 + col], v);
      }
    }
  }
}

}  // namespace

#ifndef VITFSDP_KERNELS_ONLY
std::vector<torch::Tensor> wgrad_gemm(torch::Tensor a, torch::Tensor b,
                                      bool with_bias) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
              b.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(0) == b.size(0));
  const long K = a.size(0), M = a.size(1), N = b.size(1);
  TORCH_CHECK(K % kBK == 0 && M % kBM == 0 && N % kBN == 0,
              "wgrad_gemm: needs K%64==0, M%256==0, N%256==0 (got ", K, ",",
              M, ",", N, ")");
  auto c = torch::empty({M, N}, a.options());
  torch::Tensor dbias;
  float* dbias_ptr = nullptr;
  if (with_bias) {
    dbias = torch::zeros({M}, a.options().dtype(torch::kFloat32));
    dbias_ptr = dbias.data_ptr<float>();
  }
  static bool attr_set = [] {
    hipFuncSetAttribute(reinterpret_cast<const void*>(&wgemm_atb_kernel),
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        sizeof(WgemmShared));
    return true;
  }();
  (void)attr_set;
  dim3 grid((unsigned)(M / kBM), (unsigned)(N / kBN));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(wgemm_atb_kernel, grid, dim3(kThreads),
                     sizeof(WgemmShared), stream,
                     (const short*)a.data_ptr(), (const short*)b.data_ptr(),
                     (short*)c.data_ptr(), dbias_ptr, (int)K, (int)M, (int)N);
  HIP_CHECK_LAST();
  if (with_bias) return {c, dbias};
  return {c};
}
#endif
