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
#ifndef VITFSDP_KERNELS_ONLY
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>
#endif
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
// row stride of the LDS tiles in elements: 272 keeps every tr-read lane
// address 8-byte aligned AND makes the 16-lane transpose-read groups
// bank-conflict-free (row step = 136 dwords = 8 mod 64, column slots
// 2 dwords apart: banks 8a+2b are distinct for a,b in 0..3)
constexpr int kSA = kBM + 16;
constexpr int kSB = kBN + 16;

struct WgemmShared {
  short a_tile[2][kBK][kSA];  // dY tile, row-major [k][m], double-buffered
  short b_tile[2][kBK][kSB];  // X  tile, row-major [k][n], double-buffered
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

template <bool WITH_BIAS>
__global__ __launch_bounds__(kThreads, 2) void wgemm_atb_kernel(
    const short* __restrict__ a,  // [K, M]
    const short* __restrict__ b,  // [K, N]
    short* __restrict__ c,        // [M, N]
    float* __restrict__ dbias,    // optional [M] fp32 (atomic), may be null
    int K, int M, int N) {
  HIP_DYNAMIC_SHARED(char, smem_raw)
  WgemmShared& sm = *reinterpret_cast<WgemmShared*>(smem_raw);
  const unsigned a_base0 = (unsigned)__builtin_amdgcn_groupstaticsize();
  const unsigned b_base0 = a_base0 + sizeof(sm.a_tile);
  constexpr unsigned kABufBytes = sizeof(sm.a_tile[0]);
  constexpr unsigned kBBufBytes = sizeof(sm.b_tile[0]);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int col = lane & 15;
  const int seg = lane >> 4;
  const int wm = wave >> 2;  // 0..1: wave's 128-row m strip
  const int wn = wave & 3;   // 0..3: wave's 64-col n strip

  // XCD-aware tile swizzle (guide T1): the dispatcher places block b on
  // XCD b%8, so by default neighboring tiles (which share operand
  // panels) land on different L2s and the kernel runs into the HBM
  // roofline (~7.2 TB/s effective measured without the swizzle).  The
  // remap gives each XCD a contiguous M-major run of tiles: its ~32
  // co-resident blocks then share one X panel (and neighboring dY
  // panels) in its private L2.
  int bx = blockIdx.x, by = blockIdx.y;
  {
    const int nwg = gridDim.x * gridDim.y;
    if ((nwg & 7) == 0) {
      const int bid = (int)(blockIdx.x + blockIdx.y * gridDim.x);
      const int cpx = nwg >> 3;
      const int swz = (bid & 7) * cpx + (bid >> 3);
      bx = swz % gridDim.x;
      by = swz / gridDim.x;
    }
  }
  const long m0 = (long)bx * kBM;
  const long n0 = (long)by * kBN;

  // staging coordinates: thread t covers rows (t>>5) + i*16 at the
  // constant column (t&31)*8 — pure expressions, no register arrays
  constexpr int kVecs = kBK * (kBM / 8) / kThreads;  // 4
  const int s_kr0 = tid >> 5;
  const int s_c8 = (tid & 31) * 8;
  bf16x8 a_st[kVecs], b_st[kVecs];
  auto issue_loads = [&](long k_base) {
#pragma unroll
    for (int i = 0; i < kVecs; ++i) {
      const long krow = k_base + s_kr0 + i * 16;
      a_st[i] = *reinterpret_cast<const bf16x8*>(&a[krow * M + m0 + s_c8]);
      b_st[i] = *reinterpret_cast<const bf16x8*>(&b[krow * N + n0 + s_c8]);
    }
  };
  auto write_tiles = [&](int buf) {
#pragma unroll
    for (int i = 0; i < kVecs; ++i) {
      const int kr = s_kr0 + i * 16;
      *reinterpret_cast<bf16x8*>(&sm.a_tile[buf][kr][s_c8]) = a_st[i];
      *reinterpret_cast<bf16x8*>(&sm.b_tile[buf][kr][s_c8]) = b_st[i];
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  float bias_acc[WITH_BIAS ? 8 : 1];
#pragma unroll
  for (int mi = 0; mi < (WITH_BIAS ? 8 : 1); ++mi) bias_acc[mi] = 0.f;

  const int n_ksteps = K / kBK;
  issue_loads(0);
  write_tiles(0);
  __syncthreads();

  for (int ks = 0; ks < n_ksteps; ++ks) {
    const int buf = ks & 1;
    const unsigned a_base = a_base0 + (unsigned)buf * kABufBytes;
    const unsigned b_base = b_base0 + (unsigned)buf * kBBufBytes;
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      // next K-step's loads issue at the start of the SECOND chunk:
      // their latency hides under its MFMAs and the staging registers
      // stay live for only half the loop (register budget)
      if (kc == 1 && ks + 1 < n_ksteps) issue_loads((long)(ks + 1) * kBK);
      // B fragments for this 32-k chunk (8 tr reads, reused across mi)
      u32x2 bf[4][2];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
#pragma unroll
        for (int h = 0; h < 2; ++h) {
          bf[ni][h] = tr_read(b_base, col, kc * 32 + seg * 8 + 4 * h,
                              wn * 64 + ni * 16, kSB);
        }
      }
      // A fragments stream through a 2-deep ring: af[mi+1] is issued
      // before the MFMAs of af[mi], with counted lgkmcnt waits.  The
      // waits carry the about-to-be-consumed fragments as "+v" operands
      // so the MFMAs cannot be scheduled above them (guide rule 18)
      // without fencing the whole scheduler.  (A 4-slot ring with half
      // the waits was tried: +8 live registers -> 44 VGPR spills, 2.3x
      // slower.)
      u32x2 af[2][2];
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        af[0][h] = tr_read(a_base, col, kc * 32 + seg * 8 + 4 * h,
                           wm * 128 + 0 * 16, kSA);
      }
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
        const int cur = mi & 1;
        if (mi < 7) {
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            af[cur ^ 1][h] =
                tr_read(a_base, col, kc * 32 + seg * 8 + 4 * h,
                        wm * 128 + (mi + 1) * 16, kSA);
          }
        }
        if (mi == 0) {
          // first wait also guards the B fragments (issued earlier,
          // retired in order before af[0])
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(af[cur][0]), "+v"(af[cur][1]),
                         "+v"(bf[0][0]), "+v"(bf[0][1]), "+v"(bf[1][0]),
                         "+v"(bf[1][1]), "+v"(bf[2][0]), "+v"(bf[2][1])
                       : [cnt] "i"(2)
                       : "memory");
          asm volatile("" : "+v"(bf[3][0]), "+v"(bf[3][1]));
        } else if (mi < 7) {
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(af[cur][0]), "+v"(af[cur][1])
                       : [cnt] "i"(2)
                       : "memory");
        } else {
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(af[cur][0]), "+v"(af[cur][1])
                       : [cnt] "i"(0)
                       : "memory");
        }
        union {
          u32x2 uu[2];
          bf16x8 v;
        } ac;
        ac.uu[0] = af[cur][0];
        ac.uu[1] = af[cur][1];
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          union {
            u32x2 uu[2];
            bf16x8 v;
          } bc;
          bc.uu[0] = bf[ni][0];
          bc.uu[1] = bf[ni][1];
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              ac.v, bc.v, acc[mi][ni], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
        if (WITH_BIAS && wn == 0) {
          // dbias[m] = sum_k dY[k][m]: this lane's A fragment holds 8
          // consecutive k for column m = wm*128 + mi*16 + col
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            bias_acc[mi] +=
                bf16_to_f32((unsigned short)(af[cur][h][0] & 0xffff)) +
                bf16_to_f32((unsigned short)(af[cur][h][0] >> 16)) +
                bf16_to_f32((unsigned short)(af[cur][h][1] & 0xffff)) +
                bf16_to_f32((unsigned short)(af[cur][h][1] >> 16));
          }
        }
      }
    }

    // single barrier per K-step: writes go to the OTHER buffer, so the
    // only ordering needed is "this step's writes visible before the
    // next step's reads"
    if (ks + 1 < n_ksteps) write_tiles(buf ^ 1);
    __syncthreads();
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
    if (WITH_BIAS && wn == 0) {
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

#ifdef VITFSDP_KERNELS_ONLY
template __global__ void wgemm_atb_kernel<false>(const short*, const short*,
                                                 short*, float*, int, int,
                                                 int);
#endif

}  // namespace

#ifndef VITFSDP_KERNELS_ONLY
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
    hipFuncSetAttribute(
        reinterpret_cast<const void*>(&wgemm_atb_kernel<false>),
        hipFuncAttributeMaxDynamicSharedMemorySize, sizeof(WgemmShared));
    hipFuncSetAttribute(
        reinterpret_cast<const void*>(&wgemm_atb_kernel<true>),
        hipFuncAttributeMaxDynamicSharedMemorySize, sizeof(WgemmShared));
    return true;
  }();
  (void)attr_set;
  dim3 grid((unsigned)(M / kBM), (unsigned)(N / kBN));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (with_bias) {
    hipLaunchKernelGGL(wgemm_atb_kernel<true>, grid, dim3(kThreads),
                       sizeof(WgemmShared), stream,
                       (const short*)a.data_ptr(), (const short*)b.data_ptr(),
                       (short*)c.data_ptr(), dbias_ptr, (int)K, (int)M,
                       (int)N);
  } else {
    hipLaunchKernelGGL(wgemm_atb_kernel<false>, grid, dim3(kThreads),
                       sizeof(WgemmShared), stream,
                       (const short*)a.data_ptr(), (const short*)b.data_ptr(),
                       (short*)c.data_ptr(), nullptr, (int)K, (int)M,
                       (int)N);
  }
  HIP_CHECK_LAST();
  if (with_bias) return {c, dbias};
  return {c};
}
#endif

#endif  // VITFSDP_KERNELS_ONLY
