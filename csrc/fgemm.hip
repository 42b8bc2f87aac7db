// Forward Linear GEMM for CDNA4: C[M,N] = A B^T with A = X [M, K] and
// B = W [N, K] both row-major bf16 (+ optional fused bias over N) — the
// Linear-forward shape, M = batch*tokens = 32768 for ViT-10B.
//
// This is the EASY orientation for MFMA on gfx950: the contraction dim
// K is the contiguous axis of BOTH operands, and the 16x16x32 bf16 MFMA
// wants each lane to hold 8 CONTIGUOUS k elements of its row/column —
// so both global staging and LDS fragment reads are plain vectorized
// b128 accesses (no ds_read_b64_tr_b16 hardware transposes needed,
// unlike the wgrad kernel csrc/wgemm.hip whose skeleton this reuses:
// same 256x256x64 tile, 512 threads as 2m x 4n waves of 128x64, double
// buffered LDS, next-step global loads issued under the MFMA phases).
//
// MFMA fragment layout (pinned on-device by mfma_probe,
// tests/test_gpu_kernels.py): lane l of the wave contributes row (l&15)
// of A and column (l&15) of B, k-slice (l>>4)*8 .. +8; the f32x4 result
// holds rows (l>>4)*4 .. +4 of column (l&15).

#ifndef VITFSDP_KERNELS_ONLY
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include <cstdlib>
#endif

#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int kBM = 256;
constexpr int kBN = 256;
constexpr int kThreads = 512;  // 8 waves: 2 (m) x 4 (n)

// BK is a template knob: 64 = fewer barriers, 147 KB LDS, 1 WG/CU;
// 32 = half the LDS (82 KB) so TWO workgroups co-reside per CU and one
// WG computes through the other's staging barrier.
// LDS tiles are [row][k] padded +8 elements: the row stride in dwords
// is 4*odd, so the 16 lanes of a b128 fragment-read group (consecutive
// rows, same k column) land on 16 distinct 4-dword bank slots.
template <int BK>
struct FgemmShared {
  // BK=64 stages via DirectToLds (global_load_lds 16B): lane-packed
  // rows force the PACKED stride, with bank spread done by an in-row
  // granule XOR instead of padding.  BK=32 keeps the padded layout.
  static constexpr int SK = BK == 64 ? 64 : BK + 8;
  short a_tile[2][kBM][SK];  // X tile, [m][k]
  short b_tile[2][kBN][SK];  // W tile, [n][k]
};

// XOR swizzle of the LDS column (in elements): the 16 lanes of a b128
// fragment-read group touch rows r..r+15 at one column; with any
// 16B-aligned row pad, lanes r and r+8 land on the same bank quad of
// the 32-bank LDS (measured: 4.7e8 conflict-stall cycles/dispatch,
// MfmaUtil 43.7%).  XORing the column by half the tile width for the
// upper 8 rows makes the two halves bank-disjoint.
template <int BK>
__device__ __forceinline__ int swz(int row, int col) {
  if (BK == 64) {
    // full granule spread: the 16 rows of a b128 fragment group land
    // on each bank quad exactly twice (the 2-cycle optimum for 256 B)
    return col ^ ((row & 7) * 8);
  }
  return col ^ (((row >> 3) & 1) * (BK / 2));
}

// one b128 LDS read via asm so the consuming MFMAs can be paced with
// counted lgkmcnt waits (wgemm.hip scheme; a plain dereference lets the
// compiler hoist every read and then drain with lgkmcnt(0))
__device__ __forceinline__ bf16x8 lds_read_b128(unsigned byte_addr) {
  bf16x8 v;
  asm volatile("ds_read_b128 %0, %1 offset:0" : "=v"(v) : "v"(byte_addr));
  return v;
}

// epilogue modes: 0 = plain, 1 = +bias, 2 = +bias + exact-erf GELU
// (writes the pre-activation to aux for the backward — unlike the
// hipblaslt-ext GELU_AUX_BIAS epilogue this keeps the reference's erf
// GELU numerics; ~10 extra VALU ops/element disappear under the MFMA
// phases of a compute-bound tile)
template <int EPI, int BK>
__global__ __launch_bounds__(kThreads, BK == 64 ? 1 : 2) void fgemm_abt_kernel(
    const short* __restrict__ a,     // [M, K]
    const short* __restrict__ b,     // [N, K]
    const short* __restrict__ bias,  // optional [N], may be null
    short* __restrict__ c,           // [M, N]
    short* __restrict__ aux,         // EPI==2: pre-GELU [M, N]
    int M, int N, int K) {
  constexpr bool WITH_BIAS = EPI >= 1;
  constexpr int kSK = FgemmShared<BK>::SK;
  constexpr int kChunks = BK / 32;
  HIP_DYNAMIC_SHARED(char, smem_raw)
  FgemmShared<BK>& sm = *reinterpret_cast<FgemmShared<BK>*>(smem_raw);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int col = lane & 15;  // fragment row (A) / column (B)
  const int seg = lane >> 4;  // k-slice 8*seg within the 32-k chunk
  const int wm = wave >> 2;   // 0..1: wave's 128-row m strip
  const int wn = wave & 3;    // 0..3: wave's 64-col n strip

  // XCD-aware 2D tile clustering: the dispatcher places block bid on
  // XCD bid%8, so each XCD's co-resident ~32 workgroups are
  // cid = bid>>3 consecutive.  Give each XCD a contiguous 2D
  // sub-rectangle of the tile grid and walk it in GROUP_M-wide bands:
  // the resident set then covers a GM x (32/GM) RECTANGLE, reusing A
  // panels GM-fold and B panels (32/GM)-fold in the XCD's private L2.
  // (The r1 M-major run reused only B; at 1 workgroup/CU the A streams
  // alone exceed the per-CU HBM budget — MfmaUtil measured 44%.)
  int bx = blockIdx.x, by = blockIdx.y;
  {
    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy;
    const int bid = (int)(blockIdx.x + blockIdx.y * gx);
    int sx = 2, sy = 4;  // 8 XCD sub-rectangles
    if (gy % 4 != 0) {
      if (gy % 2 == 0) { sx = 4; sy = 2; }
      else { sx = 8; sy = 1; }
    }
    if ((nwg & 7) == 0 && gx % sx == 0 && gy % sy == 0) {
      const int xcd = bid & 7, cid = bid >> 3;
      const int lw = gx / sx, lh = gy / sy;
      const int gm = (lw % 4 == 0) ? 4 : ((lw % 2 == 0) ? 2 : 1);
      const int per_band = gm * lh;
      const int band = cid / per_band, r = cid % per_band;
      bx = (xcd % sx) * lw + band * gm + (r % gm);
      by = (xcd / sx) * lh + r / gm;
    } else if ((nwg & 7) == 0) {
      const int cpx = nwg >> 3;
      const int swz = (bid & 7) * cpx + (bid >> 3);
      bx = swz % gx;
      by = swz / gx;
    }
  }
  const long m0 = (long)bx * kBM;
  const long n0 = (long)by * kBN;

  // staging:
  //  * BK=64: DirectToLds — each global_load_lds(16B) call deposits the
  //    wave's 64 lanes contiguously (1 KB = 8 packed rows), so lane l
  //    covers (row = R + (l>>3), granule l&7) and the global address is
  //    permuted per-lane to realize the bank-spread XOR.  No staging
  //    registers, no separate LDS-write phase; synced by vmcnt+barrier.
  //  * BK=32: classic register staging (padded layout).
  constexpr int kVecs = kBM * BK / 8 / kThreads;  // 4 at BK=64, 2 at BK=32
  constexpr int kPerRow = BK / 8;  // 16B vectors per tile row
  const int s_r0 = tid / kPerRow;
  const int s_c8 = (tid % kPerRow) * 8;
  constexpr int kRowsPerCall = 1024 / (BK * 2);  // 8 at BK=64
  constexpr int kCalls = kBM / (8 * kRowsPerCall);  // per wave: 4 at BK=64
  bf16x8 a_st[BK == 64 ? 1 : kVecs], b_st[BK == 64 ? 1 : kVecs];
  auto issue_dtl = [&](int buf, long k_base) {
    const int l = lane;
    const int rsub = l >> 3;
#pragma unroll
    for (int i = 0; i < kCalls; ++i) {
      const int R = (wave * kCalls + i) * kRowsPerCall;
      const int row = R + rsub;
      const int g = (l & 7) ^ (row & 7);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(
              &a[(m0 + row) * K + k_base + g * 8]),
          (__attribute__((address_space(3))) void*)&sm.a_tile[buf][R][0], 16,
          0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(
              &b[(n0 + row) * K + k_base + g * 8]),
          (__attribute__((address_space(3))) void*)&sm.b_tile[buf][R][0], 16,
          0, 0);
    }
  };
  auto issue_loads = [&](long k_base) {
    if (BK == 64) return;  // unused (DirectToLds path)
#pragma unroll
    for (int i = 0; i < kVecs && BK != 64; ++i) {
      const long row = s_r0 + i * (kThreads / kPerRow);
      a_st[i] =
          *reinterpret_cast<const bf16x8*>(&a[(m0 + row) * K + k_base + s_c8]);
      b_st[i] =
          *reinterpret_cast<const bf16x8*>(&b[(n0 + row) * K + k_base + s_c8]);
    }
  };
  auto write_tiles = [&](int buf) {
    if (BK == 64) return;
#pragma unroll
    for (int i = 0; i < kVecs && BK != 64; ++i) {
      const int row = s_r0 + i * (kThreads / kPerRow);
      const int cc = swz<BK>(row, s_c8);
      *reinterpret_cast<bf16x8*>(&sm.a_tile[buf][row][cc]) = a_st[i];
      *reinterpret_cast<bf16x8*>(&sm.b_tile[buf][row][cc]) = b_st[i];
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int n_ksteps = K / BK;
  if (BK == 64) {
    issue_dtl(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  } else {
    issue_loads(0);
    write_tiles(0);
  }
  __syncthreads();

  // byte bases of the LDS tiles for the asm-paced fragment reads
  const unsigned lds0 = (unsigned)__builtin_amdgcn_groupstaticsize();
  const unsigned a_base0 = lds0;
  const unsigned b_base0 = lds0 + (unsigned)sizeof(sm.a_tile);
  constexpr unsigned kABufBytes = sizeof(sm.a_tile[0]);
  constexpr unsigned kBBufBytes = sizeof(sm.b_tile[0]);

  for (int ks = 0; ks < n_ksteps; ++ks) {
    const int buf = ks & 1;
    const unsigned a_base = a_base0 + (unsigned)buf * kABufBytes;
    const unsigned b_base = b_base0 + (unsigned)buf * kBBufBytes;
#pragma unroll
    for (int kc = 0; kc < kChunks; ++kc) {
      // next K-step's loads issue early: DirectToLds at the FIRST
      // chunk (no registers held), register staging at the last
      if (BK == 64) {
        if (kc == 0 && ks + 1 < n_ksteps) issue_dtl(buf ^ 1, (long)(ks + 1) * BK);
      } else if (kc == kChunks - 1 && ks + 1 < n_ksteps) {
        issue_loads((long)(ks + 1) * BK);
      }
      const int kcol = kc * 32 + seg * 8;
      // B fragments for this chunk (4 b128 reads, reused across all 8
      // m fragments), then the A fragments stream through a 2-deep
      // ring with counted lgkmcnt waits so MFMAs start as soon as
      // their own fragment retires (wgemm.hip pacing)
      bf16x8 bf[4];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int br = wn * 64 + ni * 16 + col;
        bf[ni] = lds_read_b128(
            b_base + 2u * ((unsigned)(br * kSK + swz<BK>(br, kcol))));
      }
      // all 8 A fragments issued up-front: one deep DS pipeline, a
      // single counted wait per mi instead of a 2-deep ring (the ring
      // measured no better than plain; this removes the per-mi
      // round-trips entirely at +32 VGPRs)
      bf16x8 af[8];
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
        const int ar = wm * 128 + mi * 16 + col;
        af[mi] = lds_read_b128(
            a_base + 2u * ((unsigned)(ar * kSK + swz<BK>(ar, kcol))));
      }
      asm volatile("s_waitcnt lgkmcnt(0)"
                   : "+v"(af[0]), "+v"(af[1]), "+v"(af[2]), "+v"(af[3]),
                     "+v"(af[4]), "+v"(af[5]), "+v"(af[6]), "+v"(af[7]),
                     "+v"(bf[0]), "+v"(bf[1]), "+v"(bf[2]), "+v"(bf[3])
                   :
                   : "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    if (BK == 64) {
      if (ks + 1 < n_ksteps)
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    } else if (ks + 1 < n_ksteps) {
      write_tiles(buf ^ 1);
    }
    __syncthreads();
  }

  // epilogue: this lane owns rows m = ... + seg*4 + r of column n
  float bias_v[4];
  if (WITH_BIAS) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      bias_v[ni] = bf16_to_f32(
          (unsigned short)bias[n0 + wn * 64 + ni * 16 + col]);
    }
  }
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long m = m0 + wm * 128 + mi * 16 + seg * 4 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const long n = n0 + wn * 64 + ni * 16 + col;
        float v = acc[mi][ni][r];
        if (WITH_BIAS) v += bias_v[ni];
        if (EPI == 2) {
          aux[m * N + n] = (short)f32_to_bf16(v);
          // exact erf GELU (reference numerics, torch F.gelu default)
          v = 0.5f * v * (1.f + erff(v * 0.70710678118654752f));
        }
        c[m * N + n] = (short)f32_to_bf16(v);
      }
    }
  }
}

typedef __attribute__((ext_vector_type(16))) float f32x16;

// Wide-MFMA variant: mfma_f32_32x32x16_bf16 halves the MFMA instruction
// count per FLOP (the 16x16 kernel measured VALUBusy 27% competing with
// MFMA issue at MfmaUtil 53%).  BK=64 + DirectToLds staging + the same
// clustered mapping; fragment layout pinned by mfma32_probe
// (tests/test_gpu_kernels.py::test_mfma32_probe_layout).
template <int EPI>
__global__ __launch_bounds__(kThreads, 1) void fgemm_abt_mi32_kernel(
    const short* __restrict__ a, const short* __restrict__ b,
    const short* __restrict__ bias, short* __restrict__ c,
    short* __restrict__ aux, int M, int N, int K) {
  constexpr bool WITH_BIAS = EPI >= 1;
  constexpr int BK = 64;
  HIP_DYNAMIC_SHARED(char, smem_raw)
  FgemmShared<BK>& sm = *reinterpret_cast<FgemmShared<BK>*>(smem_raw);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31;  // fragment row (A) / column (B)
  const int kseg = lane >> 5;  // k-slice 8*kseg within a 16-k chunk
  const int wm = wave >> 2;   // 0..1: wave's 128-row m strip
  const int wn = wave & 3;    // 0..3: wave's 64-col n strip

  int bx = blockIdx.x, by = blockIdx.y;
  {
    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy;
    const int bid = (int)(blockIdx.x + blockIdx.y * gx);
    int sx = 2, sy = 4;
    if (gy % 4 != 0) {
      if (gy % 2 == 0) { sx = 4; sy = 2; }
      else { sx = 8; sy = 1; }
    }
    if ((nwg & 7) == 0 && gx % sx == 0 && gy % sy == 0) {
      const int xcd = bid & 7, cid = bid >> 3;
      const int lw = gx / sx, lh = gy / sy;
      const int gm = (lw % 4 == 0) ? 4 : ((lw % 2 == 0) ? 2 : 1);
      const int per_band = gm * lh;
      const int band = cid / per_band, r = cid % per_band;
      bx = (xcd % sx) * lw + band * gm + (r % gm);
      by = (xcd / sx) * lh + r / gm;
    } else if ((nwg & 7) == 0) {
      const int cpx = nwg >> 3;
      const int swzb = (bid & 7) * cpx + (bid >> 3);
      bx = swzb % gx;
      by = swzb / gx;
    }
  }
  const long m0 = (long)bx * kBM;
  const long n0 = (long)by * kBN;

  // DirectToLds staging (same scheme as the 16x16 kernel)
  auto issue_dtl = [&](int buf, long k_base) {
    const int rsub = lane >> 3;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int R = (wave * 4 + i) * 8;
      const int row = R + rsub;
      const int g = (lane & 7) ^ (row & 7);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(
              &a[(m0 + row) * K + k_base + g * 8]),
          (__attribute__((address_space(3))) void*)&sm.a_tile[buf][R][0], 16,
          0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(
              &b[(n0 + row) * K + k_base + g * 8]),
          (__attribute__((address_space(3))) void*)&sm.b_tile[buf][R][0], 16,
          0, 0);
    }
  };

  f32x16 acc[4][2];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = f32x16{};

  const int n_ksteps = K / BK;
  issue_dtl(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int ks = 0; ks < n_ksteps; ++ks) {
    const int buf = ks & 1;
    if (ks + 1 < n_ksteps) issue_dtl(buf ^ 1, (long)(ks + 1) * BK);
#pragma unroll
    for (int kc = 0; kc < 4; ++kc) {  // 4 chunks of k=16
      const int kcol = kc * 16 + kseg * 8;
      bf16x8 bf[2];
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int br = wn * 64 + ni * 32 + l31;
        bf[ni] = *reinterpret_cast<const bf16x8*>(
            &sm.b_tile[buf][br][swz<BK>(br, kcol)]);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int ar = wm * 128 + mi * 32 + l31;
        const bf16x8 af = *reinterpret_cast<const bf16x8*>(
            &sm.a_tile[buf][ar][swz<BK>(ar, kcol)]);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, bf[ni], acc[mi][ni], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    if (ks + 1 < n_ksteps)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // epilogue: lane holds col l31 of rows (lane>>5)*4 + g*8 + r
  float bias_v[2];
  if (WITH_BIAS) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      bias_v[ni] = bf16_to_f32(
          (unsigned short)bias[n0 + wn * 64 + ni * 32 + l31]);
    }
  }
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int g = 0; g < 4; ++g) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wm * 128 + mi * 32 + kseg * 4 + g * 8 + r;
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const long n = n0 + wn * 64 + ni * 32 + l31;
          float v = acc[mi][ni][g * 4 + r];
          if (WITH_BIAS) v += bias_v[ni];
          if (EPI == 2) {
            aux[m * N + n] = (short)f32_to_bf16(v);
            v = 0.5f * v * (1.f + erff(v * 0.70710678118654752f));
          }
          c[m * N + n] = (short)f32_to_bf16(v);
        }
      }
    }
  }
}

#ifdef VITFSDP_KERNELS_ONLY
template __global__ void fgemm_abt_kernel<0, 64>(const short*, const short*,
                                                 const short*, short*, short*,
                                                 int, int, int);
template __global__ void fgemm_abt_kernel<1, 64>(const short*, const short*,
                                                 const short*, short*, short*,
                                                 int, int, int);
template __global__ void fgemm_abt_kernel<2, 64>(const short*, const short*,
                                                 const short*, short*, short*,
                                                 int, int, int);
template __global__ void fgemm_abt_kernel<0, 32>(const short*, const short*,
                                                 const short*, short*, short*,
                                                 int, int, int);
template __global__ void fgemm_abt_mi32_kernel<0>(const short*, const short*,
                                                  const short*, short*,
                                                  short*, int, int, int);
#endif

}  // namespace

#ifndef VITFSDP_KERNELS_ONLY
namespace {

// runtime BK selection (A/B knob): 64 (default) or 32 (2 WG/CU)
int fgemm_bk() {
  static int bk = [] {
    const char* e = std::getenv("VITFSDP_FGEMM_BK");
    return (e != nullptr && std::atoi(e) == 32) ? 32 : 64;
  }();
  return bk;
}

// MFMA shape selection via VITFSDP_FGEMM_MI: 16 (mfma_16x16x32,
// default — measured 1109-1168 TF/s) or 32 (mfma_32x32x16 — halves the
// MFMA instruction count but measured 1048-1082: the longer-latency
// wide MFMA with only 2 independent n-chains per m-tile loses more to
// dependency stalls than it gains in issue slots)
int fgemm_mi() {
  static int mi = [] {
    const char* e = std::getenv("VITFSDP_FGEMM_MI");
    return (e != nullptr && std::atoi(e) == 32) ? 32 : 16;
  }();
  return mi;
}

template <int EPI, int BK>
void launch_fgemm(const short* x, const short* w, const short* bias, short* c,
                  short* aux, long M, long N, long K, hipStream_t stream) {
  static bool attr_set = [] {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&fgemm_abt_kernel<EPI, BK>),
        hipFuncAttributeMaxDynamicSharedMemorySize, sizeof(FgemmShared<BK>));
    return true;
  }();
  (void)attr_set;
  dim3 grid((unsigned)(M / kBM), (unsigned)(N / kBN));
  hipLaunchKernelGGL((fgemm_abt_kernel<EPI, BK>), grid, dim3(kThreads),
                     sizeof(FgemmShared<BK>), stream, x, w, bias, c, aux,
                     (int)M, (int)N, (int)K);
}

template <int EPI>
void launch_fgemm_mi32(const short* x, const short* w, const short* bias,
                       short* c, short* aux, long M, long N, long K,
                       hipStream_t stream) {
  static bool attr_set = [] {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&fgemm_abt_mi32_kernel<EPI>),
        hipFuncAttributeMaxDynamicSharedMemorySize, sizeof(FgemmShared<64>));
    return true;
  }();
  (void)attr_set;
  dim3 grid((unsigned)(M / kBM), (unsigned)(N / kBN));
  hipLaunchKernelGGL((fgemm_abt_mi32_kernel<EPI>), grid, dim3(kThreads),
                     sizeof(FgemmShared<64>), stream, x, w, bias, c, aux,
                     (int)M, (int)N, (int)K);
}

template <int EPI>
void launch_fgemm_bk(const short* x, const short* w, const short* bias,
                     short* c, short* aux, long M, long N, long K,
                     hipStream_t stream) {
  if (fgemm_bk() == 32 && K % 32 == 0)
    launch_fgemm<EPI, 32>(x, w, bias, c, aux, M, N, K, stream);
  else if (fgemm_mi() == 32)
    launch_fgemm_mi32<EPI>(x, w, bias, c, aux, M, N, K, stream);
  else
    launch_fgemm<EPI, 64>(x, w, bias, c, aux, M, N, K, stream);
}

}  // namespace

torch::Tensor fwd_gemm(torch::Tensor x, torch::Tensor w,
                       c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous(),
              "fwd_gemm: contiguous CUDA tensors required");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16, "fwd_gemm: bf16 only");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1),
              "fwd_gemm: x[M,K], w[N,K] expected");
  const long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M % kBM == 0 && N % kBN == 0 && K % 64 == 0,
              "fwd_gemm: needs M%256==0, N%256==0, K%64==0 (got ", M, ",", N,
              ",", K, ")");
  auto c = torch::empty({M, N}, x.options());
  const short* bias_ptr = nullptr;
  torch::Tensor bias_c;
  if (bias.has_value()) {
    bias_c = bias->contiguous();
    TORCH_CHECK(bias_c.scalar_type() == torch::kBFloat16 &&
                bias_c.numel() == N, "fwd_gemm: bias must be bf16 [N]");
    bias_ptr = (const short*)bias_c.data_ptr();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  if (bias_ptr) {
    launch_fgemm_bk<1>((const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       bias_ptr, (short*)c.data_ptr(), nullptr, M, N, K,
                       stream);
  } else {
    launch_fgemm_bk<0>((const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       nullptr, (short*)c.data_ptr(), nullptr, M, N, K,
                       stream);
  }
  HIP_CHECK_LAST();
  return c;
}

// (gelu(x @ w^T + bias), pre-activation): the erf-exact fused MLP fc1
// forward — our answer to hipblaslt-ext GELU_AUX_BIAS without the tanh
// approximation (kernel K6 fusion).
std::vector<torch::Tensor> fwd_gemm_gelu(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous() &&
              bias.is_contiguous(), "fwd_gemm_gelu: contiguous CUDA only");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16 &&
              bias.scalar_type() == torch::kBFloat16,
              "fwd_gemm_gelu: bf16 only");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(1) &&
              bias.numel() == w.size(0), "fwd_gemm_gelu: x[M,K], w[N,K]");
  const long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M % kBM == 0 && N % kBN == 0 && K % 64 == 0,
              "fwd_gemm_gelu: needs M%256==0, N%256==0, K%64==0");
  auto c = torch::empty({M, N}, x.options());
  auto aux = torch::empty({M, N}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  launch_fgemm_bk<2>((const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     (const short*)bias.data_ptr(), (short*)c.data_ptr(),
                     (short*)aux.data_ptr(), M, N, K, stream);
  HIP_CHECK_LAST();
  return {c, aux};
}
#endif  // VITFSDP_KERNELS_ONLY
