// Flash-style fused multi-head attention for CDNA4 (kernel K4,
// SURVEY.md §2D): per-head softmax(Q K^T * scale) V with online softmax,
// O(T) memory, for the ViT-10B head_dim of 160 (= 5120/32, reference
// run_vit_training.py:341-342) — off the usual 64/128 fast paths, so the
// tiling is D-generic over 32-wide MFMA K-chunks (160 = 5 x 32).
//
// head_dim is a TEMPLATE parameter: every fragment/accumulator loop is
// compile-time bounded so the f32x4 MFMA accumulators stay in VGPRs — a
// runtime-indexed ext_vector array silently spills to scratch, which
// measured ~300x slower (guide §5.4 rule 20; seen in the first profile,
// profiles/).
//
// Structure (forward):
//   grid = (B*H, ceil(T/QTile)); block = 256 threads = 4 waves; K/V
//   tiles double-buffered in LDS with T14 split staging (next tile's
//   HBM loads issue before this tile's MFMA phases), one barrier/tile.
//   Per 32-key tile:
//     - SWAPPED S^T = K Q^T via __builtin_amdgcn_mfma_f32_16x16x32_bf16
//       (C-layout rows = k, cols = q: each lane owns ONE q row),
//     - fully in-register online softmax (7 fmax + 2 cross-seg shuffles
//       per row; state = two scalars per lane),
//     - P re-enters the MFMA A layout via 4 packed bf16-pair shuffles
//       (no LDS round-trip, no lgkmcnt drain),
//     - O += P V with B fragments gathered from the ROW-major V tile
//       by ds_read_b64_tr_b16 hardware transpose reads (2-deep ring).
//   Epilogue: O /= rowsum, store bf16, write LSE (fp32) for backward.
//
// Backward: fully fused (FlashAttention-2 style), two MFMA kernels —
// fmha_bwd_dq (q-tile ownership) and fmha_bwd_dkv (k-tile ownership),
// both recompute P from the forward LSE; Delta = rowsum(dO*O) is its
// own reduction kernel.  No batched GEMM library calls (the hipBLASLt
// composition survives as a debug path, VITFSDP_FMHA_BWD=compose).
//
// MFMA fragment layout (verified on-device by mfma_probe,
// tests/test_gpu_kernels.py::test_mfma_probe_layout):
//   A[i][k]: lane l holds a[j] = A[l&15][(l>>4)*8 + j]
//   B[k][j]: lane l holds b[j] = B[(l>>4)*8 + j][l&15]
//   C[i][j]: lane l holds c[r] = C[(l>>4)*4 + r][l&15]

// Compile with -DVITFSDP_KERNELS_ONLY to build just the device kernels
// (no torch headers) — used by csrc/tools/check_resources.sh for
// -Rpass-analysis=kernel-resource-usage without the extension build.
#ifndef VITFSDP_KERNELS_ONLY
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include <cstdlib>
#include <cstring>
#endif

#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2;

constexpr int kBlockThreads = 256;
constexpr int kKTile = 32;  // k rows per inner iteration
constexpr int kMaxD = 192;  // supports head_dim up to 192 (10B uses 160)

template <int D>
struct FmhaShapes {
  // 16-row q sub-tiles per wave: 2 where the register budget allows >=2
  // waves/SIMD occupancy; 1 for large head_dim (at D=160, QSub=2 costs
  // 228 VGPR + 80 AGPR -> 1 wave/SIMD, measured slower than QSub=1)
  static constexpr int QSub = (D <= 96) ? 2 : 1;
  static constexpr int QTile = 64 * QSub;  // q rows per workgroup
  static constexpr int DP = ((D + 31) / 32) * 32;  // zero-padded for QK
  static constexpr int NKC = DP / 32;              // QK^T K-chunks
  static constexpr int NC = D / 16;                // PV / O column chunks
  // LDS row strides in elements, padded so the 16-lane ds_read_b128
  // column groups land on distinct banks (stride*2B stays 16B-aligned)
  static constexpr int KStride = DP + 8;
  static constexpr int PStride = kKTile + 8;  // 40
  static constexpr int VRow = D + 16;  // row-major V: 8B-aligned tr
                                       // lanes, conflict-free groups
  struct Shared {
    short k_tile[2][kKTile][KStride];  // double-buffered
    short v_tile[2][kKTile][VRow];     // ROW-major [k][d], dbuf: PV reads
                                       // it via ds_read_b64_tr_b16
  };
  // The backward kernels keep ONLY row-major tiles: the operands that
  // need K-contiguous fragments (dQ's K, dKV's Q/dO) are read with
  // ds_read_b64_tr_b16 hardware transpose reads, so the element-wise
  // transposed copies (and their bank-conflict swizzles) are gone.
  struct SharedDQ {
    short k_tile[kKTile][KStride];  // row-major (S^T + dQ operands)
    short v_tile[kKTile][KStride];  // row-major (dP^T operand)
  };
  struct SharedDKV {
    short q_tile[32][KStride];    // row-major (S^T + dK operands)
    short do_tile[32][KStride];   // row-major (dP^T + dV operands)
    short pt_tile[4][16][PStride];   // P^T re-layout per wave
    short dst_tile[4][16][PStride];  // dS^T re-layout per wave
  };
};


// grid = (B*H, ceil(T/kQTile)): bh on x so a head's q-tiles share the
// XCD whose L2 already holds its K/V (blockIdx linearization round-
// robins x across XCDs; y strides keep bh%8 constant -> same XCD)
// Strides (in elements, d innermost contiguous): the q/k/v pointers
// share one stride set (they are views of the fused qkv projection
// [B,T,3,H,D]); o has its own (it is written [B,T,H,D] = [B,T,E], so the
// attention module needs no permute/contiguous copies on either side).
struct QkvStrides {
  long qb, qh, qt;  // q/k/v batch, head, token strides
  long ob, oh, ot;  // o (and dO) strides
};

// Attention-dropout mask: stateless integer hash of (seed, global q row
// incl. bh, k column) so forward and both backward kernels regenerate
// the identical mask with no stored bits.  keep iff hash >= threshold
// (threshold = p * 2^32); integer-exact and reproduced bit-for-bit by
// the python reference in tests/test_gpu_kernels.py.
__device__ __forceinline__ unsigned dropout_hash(unsigned seed, unsigned qg,
                                                 unsigned kg) {
  unsigned x = seed ^ (qg * 0x9E3779B9u) ^ (kg * 0x85EBCA6Bu);
  x ^= x >> 16;
  x *= 0x7FEB352Du;
  x ^= x >> 15;
  x *= 0x846CA68Bu;
  x ^= x >> 16;
  return x;
}

template <int D, bool DROP = false>
// min 3 waves/SIMD: the double-buffered staging otherwise lands at
// 182 regs -> 2 waves; forcing 3 costs a few prologue spills at most
__global__ __launch_bounds__(kBlockThreads, 3) void fmha_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse_out, int T, int H, QkvStrides st, float scale,
    unsigned drop_thresh = 0, float drop_rscale = 1.f, unsigned seed = 0) {
  using S = FmhaShapes<D>;
  constexpr int DP = S::DP, NKC = S::NKC, NC = S::NC;
  constexpr int kQSub = S::QSub;
  constexpr int kQTile = S::QTile;
  // loads per thread per K/V tile (16-byte vectors)
  constexpr int kKVecs = (kKTile * (DP / 8) + kBlockThreads - 1) / kBlockThreads;
  constexpr int kVVecs = (kKTile * (D / 8) + kBlockThreads - 1) / kBlockThreads;
  HIP_DYNAMIC_SHARED(char, smem_raw)
  typename S::Shared& sm = *reinterpret_cast<typename S::Shared*>(smem_raw);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;  // fragment column group
  const int seg = lane >> 4;  // fragment k/row segment (0..3)
  const long bh = blockIdx.x;
  const int q_base = blockIdx.y * kQTile;
  // this wave's first q row, per sub-tile qs: q_row0 + 16*qs
  const int q_row0 = q_base + wave * (16 * kQSub);

  const int b_idx = (int)(bh / H), h_idx = (int)(bh % H);
  const long qkv_off = (long)b_idx * st.qb + (long)h_idx * st.qh;
  const long o_off = (long)b_idx * st.ob + (long)h_idx * st.oh;

  // ---- hoisted staging coordinates (reused every K/V tile) ----
  int k_kr[kKVecs], k_dc[kKVecs];
#pragma unroll
  for (int i = 0; i < kKVecs; ++i) {
    const int idx = (int)threadIdx.x + i * kBlockThreads;
    k_kr[i] = idx / (DP / 8);
    k_dc[i] = (idx % (DP / 8)) * 8;
  }
  int v_kr[kVVecs], v_dc[kVVecs];
#pragma unroll
  for (int i = 0; i < kVVecs; ++i) {
    const int idx = (int)threadIdx.x + i * kBlockThreads;
    v_kr[i] = idx / (D / 8);
    v_dc[i] = (idx % (D / 8)) * 8;
  }

  // ---- load Q fragments to registers: sub-tile qs, chunk kc, elem j ->
  // Q[q_row0 + 16*qs + col][seg*8 + j + 32*kc] (zero-padded beyond D/T) --
  bf16x8 q_frag[kQSub][NKC];
#pragma unroll
  for (int qs = 0; qs < kQSub; ++qs) {
    const int q_row = q_row0 + 16 * qs + col;
    const bool valid = q_row < T;
    const long base = qkv_off + (long)q_row * st.qt;
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      const int d0 = kc * 32 + seg * 8;
      if (valid && d0 + 8 <= D) {
        // contiguous 8 elements: one 16-B load instead of 8 scalars
        q_frag[qs][kc] = *reinterpret_cast<const bf16x8*>(&q[base + d0]);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          q_frag[qs][kc][j] =
              (valid && d0 + j < D) ? q[base + d0 + j] : (short)0;
        }
      }
    }
  }

  // ---- online softmax state: ONE q row per lane (q = col) ----
  float m_run[kQSub], l_run[kQSub];
  f32x4 o_acc[kQSub][NC];
#pragma unroll
  for (int qs = 0; qs < kQSub; ++qs) {
    m_run[qs] = -INFINITY;
    l_run[qs] = 0.f;
#pragma unroll
    for (int c = 0; c < NC; ++c) o_acc[qs][c] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  // ---- staging helpers (double-buffered K/V, T14 split: the NEXT
  // tile's global loads are issued before this tile's compute so HBM
  // latency hides under the MFMA phases; the LDS writes land after) ----
  // K and V staging share one register block: their live ranges are
  // disjoint (K: issue -> write before PV; V: issue -> write after PV)
  constexpr int kStageVecs = kKVecs > kVVecs ? kKVecs : kVVecs;
  bf16x8 stage[kStageVecs];
  auto issue_k_loads = [&](int k_base) {
#pragma unroll
    for (int i = 0; i < kKVecs; ++i) {
      bf16x8 val = {0, 0, 0, 0, 0, 0, 0, 0};
      const int k_row = k_base + k_kr[i];
      if (k_kr[i] < kKTile && k_row < T && k_dc[i] < D) {
        val = *reinterpret_cast<const bf16x8*>(
            &k[qkv_off + (long)k_row * st.qt + k_dc[i]]);
      }
      stage[i] = val;
    }
  };
  auto write_k_tile = [&](int buf) {
#pragma unroll
    for (int i = 0; i < kKVecs; ++i) {
      if (k_kr[i] < kKTile) {
        *reinterpret_cast<bf16x8*>(&sm.k_tile[buf][k_kr[i]][k_dc[i]]) =
            stage[i];
      }
    }
  };
  auto issue_v_loads = [&](int k_base) {
#pragma unroll
    for (int i = 0; i < kVVecs; ++i) {
      bf16x8 val = {0, 0, 0, 0, 0, 0, 0, 0};
      const int k_row = k_base + v_kr[i];
      if (v_kr[i] < kKTile && k_row < T) {
        val = *reinterpret_cast<const bf16x8*>(
            &v[qkv_off + (long)k_row * st.qt + v_dc[i]]);
      }
      stage[i] = val;
    }
  };
  auto write_v_tile = [&](int buf) {
#pragma unroll
    for (int i = 0; i < kVVecs; ++i) {
      if (v_kr[i] < kKTile) {
        *reinterpret_cast<bf16x8*>(&sm.v_tile[buf][v_kr[i]][v_dc[i]]) =
            stage[i];
      }
    }
  };

  const int n_ktiles = (T + kKTile - 1) / kKTile;
  // prologue: stage tile 0 into buffer 0
  issue_k_loads(0);
  write_k_tile(0);
  issue_v_loads(0);
  write_v_tile(0);
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < n_ktiles; ++kt) {
    const int k_base = kt * kKTile;
    const bool has_next = kt + 1 < n_ktiles;

    // issue next K loads; their latency hides under this tile's QK^T
    if (has_next) issue_k_loads(k_base + kKTile);

    // ---- per q sub-tile: S^T = scale*(K Q^T), in-register softmax ----
    // Swapped operands (guide §B attn): mfma(A=K, B=Q) puts the scores
    // in C layout rows = k, cols = q — each lane holds the 8 scores of
    // ONE q row (q = col), so the row reduce is 7 in-register ops plus
    // two cross-seg shuffles, the softmax state is two scalars per
    // lane, and P returns to the MFMA A layout with four packed
    // shuffles instead of an LDS round-trip + lgkmcnt(0) drain.
    // Q's register fragments serve both orders: its A-layout fragment
    // (lane: Q[col][seg*8+j]) IS its B-layout fragment for the swap.
#pragma unroll
    for (int qs = 0; qs < kQSub; ++qs) {
      f32x4 st_frag[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int kc = 0; kc < NKC; ++kc) {
          // A fragment: A[col][seg*8+j] = K[kk*16+col][kc*32+seg*8+j]
          bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
              &sm.k_tile[cur][kk * 16 + col][kc * 32 + seg * 8]);
          st_frag[kk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, q_frag[qs][kc], st_frag[kk], 0, 0, 0);
        }
      }

      // lane's 8 scores: q = col, k = k_base + 16*kk + seg*4 + r
      float sc[2][4];
      float my_max = -INFINITY;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float sv = st_frag[kk][r] * scale;
          if (k_base + 16 * kk + seg * 4 + r >= T) sv = -INFINITY;
          sc[kk][r] = sv;
          my_max = fmaxf(my_max, sv);
        }
      }
      my_max = fmaxf(my_max, __shfl_xor(my_max, 16));
      my_max = fmaxf(my_max, __shfl_xor(my_max, 32));
      const float m_new = fmaxf(m_run[qs], my_max);
      const float alpha = __expf(m_run[qs] - m_new);
      m_run[qs] = m_new;

      float p[2][4];
      float row_sum = 0.f;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float pv =
              (sc[kk][r] == -INFINITY) ? 0.f : __expf(sc[kk][r] - m_new);
          p[kk][r] = pv;
          row_sum += pv;
        }
      }
      row_sum += __shfl_xor(row_sum, 16);
      row_sum += __shfl_xor(row_sum, 32);
      l_run[qs] = l_run[qs] * alpha + row_sum;

      if (DROP) {
        // dropout on P (post-softmax, reference nn.Dropout semantics):
        // the row sum above stays UNMASKED (it is the softmax
        // normalizer); the PV path gets the masked, 1/(1-p)-scaled P
        const unsigned qg = (unsigned)(bh * T + q_row0 + 16 * qs + col);
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const unsigned kg = (unsigned)(k_base + 16 * kk + seg * 4 + r);
            p[kk][r] = dropout_hash(seed, qg, kg) >= drop_thresh
                           ? p[kk][r] * drop_rscale
                           : 0.f;
          }
        }
      }

      // pack P to bf16 pairs (consecutive k): pk[kk][rr] holds
      // (k = 16*kk + seg*4 + 2*rr, +1) for q = col
      unsigned pk[2][2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
          pk[kk][rr] = (unsigned)f32_to_bf16(p[kk][2 * rr]) |
                       ((unsigned)f32_to_bf16(p[kk][2 * rr + 1]) << 16);
        }
      }
      // redistribute to the MFMA A layout: target elem pair j2 wants
      // k = seg*8 + 2*j2 of its own q=col, which lives in lane
      // col + 16*((seg&1)*2 + (j2>>1)), slot (kk=seg>>1, rr=j2&1).
      // A source lane serves targets from BOTH kk halves, so shuffle
      // both kk slots and let each target select by its own seg.
      const int kk_s = seg >> 1;
      unsigned pa_u[4];
#pragma unroll
      for (int j2 = 0; j2 < 4; ++j2) {
        const int src = col + 16 * ((seg & 1) * 2 + (j2 >> 1));
        const unsigned lo = (unsigned)__shfl((int)pk[0][j2 & 1], src);
        const unsigned hi = (unsigned)__shfl((int)pk[1][j2 & 1], src);
        pa_u[j2] = kk_s ? hi : lo;
      }
      bf16x8 pa;
      {
        union {
          unsigned u[4];
          bf16x8 v;
        } cvt;
        cvt.u[0] = pa_u[0];
        cvt.u[1] = pa_u[1];
        cvt.u[2] = pa_u[2];
        cvt.u[3] = pa_u[3];
        pa = cvt.v;
      }

      // rescale O: its C-layout rows are q = seg*4 + r, whose alpha
      // lives in lane (seg*4 + r) of the column group
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float alpha_o = __shfl(alpha, seg * 4 + r);
#pragma unroll
        for (int c = 0; c < NC; ++c) o_acc[qs][c][r] *= alpha_o;
      }

      // K(t+1) in flight and QK^T(t) done for the LAST sub-tile: write
      // it, then issue V(t+1) loads to hide under PV
      if (qs == kQSub - 1 && has_next) {
        write_k_tile(cur ^ 1);
        issue_v_loads(k_base + kKTile);
      }

      // ---- O += P V : NC chunks of 16 output columns ----
      // B fragments come from the ROW-major V tile via
      // ds_read_b64_tr_b16 (2 transpose reads per 16-column chunk,
      // 2-deep ring with counted waits): the element-wise transposed
      // store and its bank-conflict swizzle are gone entirely.
      const unsigned v_lds_base =
          (unsigned)__builtin_amdgcn_groupstaticsize() +
          (unsigned)(offsetof(typename S::Shared, v_tile)) +
          (unsigned)cur * (unsigned)sizeof(sm.v_tile[0]);
      auto v_tr = [&](int c, int h) -> u32x2 {
        const unsigned addr =
            v_lds_base +
            2u * ((unsigned)((seg * 8 + 4 * h + (col >> 2)) * S::VRow +
                             c * 16 + (col & 3) * 4));
        u32x2 r;
        asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
                     : "=v"(r)
                     : "v"(addr));
        return r;
      };
      u32x2 vf[2][2];
#pragma unroll
      for (int h = 0; h < 2; ++h) vf[0][h] = v_tr(0, h);
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        const int curc = c & 1;
        if (c + 1 < NC) {
#pragma unroll
          for (int h = 0; h < 2; ++h) vf[curc ^ 1][h] = v_tr(c + 1, h);
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(vf[curc][0]), "+v"(vf[curc][1])
                       : [cnt] "i"(2)
                       : "memory");
        } else {
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(vf[curc][0]), "+v"(vf[curc][1])
                       : [cnt] "i"(0)
                       : "memory");
        }
        union {
          u32x2 uu[2];
          bf16x8 v;
        } vc;
        vc.uu[0] = vf[curc][0];
        vc.uu[1] = vf[curc][1];
        o_acc[qs][c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, vc.v, o_acc[qs][c], 0, 0, 0);
      }
    }

    if (has_next) write_v_tile(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: normalize, store O and LSE ----
#pragma unroll
  for (int qs = 0; qs < kQSub; ++qs) {
    // this lane's own row stats (q = col) feed the LSE store; the O
    // rows it holds (q = seg*4 + r) read their l from the owning lane
    if (seg == 0) {
      const int q_row = q_row0 + 16 * qs + col;
      if (q_row < T) {
        lse_out[bh * T + q_row] = m_run[qs] + __logf(l_run[qs]);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float l_row = __shfl(l_run[qs], seg * 4 + r);
      const int q_row = q_row0 + 16 * qs + seg * 4 + r;
      if (q_row >= T) continue;
      const float inv_l = (l_row > 0.f) ? 1.f / l_row : 0.f;
      const long out_base = o_off + (long)q_row * st.ot;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        o[out_base + c * 16 + col] =
            (short)f32_to_bf16(o_acc[qs][c][r] * inv_l);
      }
    }
  }
}

template <int D, bool DROP = false>
__global__ __launch_bounds__(kBlockThreads, 3) void fmha_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq, int T, int H, QkvStrides st, float scale,
    unsigned drop_thresh = 0, float drop_rscale = 1.f, unsigned seed = 0) {
  using S = FmhaShapes<D>;
  constexpr int DP = S::DP, NKC = S::NKC, NC = S::NC;
  constexpr int kQSub = S::QSub;
  constexpr int kQTile = S::QTile;
  HIP_DYNAMIC_SHARED(char, smem_raw)
  typename S::SharedDQ& sm =
      *reinterpret_cast<typename S::SharedDQ*>(smem_raw);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int seg = lane >> 4;
  const long bh = blockIdx.x;
  const int q_base = blockIdx.y * kQTile;
  const int q_row0 = q_base + wave * (16 * kQSub);
  const int b_idx = (int)(bh / H), h_idx = (int)(bh % H);
  const long qkv_off = (long)b_idx * st.qb + (long)h_idx * st.qh;
  const long o_off = (long)b_idx * st.ob + (long)h_idx * st.oh;

  // Q and dO register fragments: the same registers serve as the B
  // operands of the swapped S^T / dP^T products (see the forward)
  bf16x8 q_frag[kQSub][NKC], do_frag[kQSub][NKC];
  float lse_q[kQSub], delta_q[kQSub];
#pragma unroll
  for (int qs = 0; qs < kQSub; ++qs) {
    const int q_row = q_row0 + 16 * qs + col;
    const bool valid = q_row < T;
    const long base = qkv_off + (long)q_row * st.qt;
    const long dbase = o_off + (long)q_row * st.ot;
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      const int d0 = kc * 32 + seg * 8;
      if (valid && d0 + 8 <= D) {
        q_frag[qs][kc] = *reinterpret_cast<const bf16x8*>(&q[base + d0]);
        do_frag[qs][kc] =
            *reinterpret_cast<const bf16x8*>(&dout[dbase + d0]);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const bool ok = valid && d0 + j < D;
          q_frag[qs][kc][j] = ok ? q[base + d0 + j] : (short)0;
          do_frag[qs][kc][j] = ok ? dout[dbase + d0 + j] : (short)0;
        }
      }
    }
    // per-lane row stats: this lane's q row is q = col (swapped layout)
    lse_q[qs] = valid ? lse[bh * T + q_row] : 0.f;
    delta_q[qs] = valid ? delta[bh * T + q_row] : 0.f;
  }

  f32x4 dq_acc[kQSub][NC];
#pragma unroll
  for (int qs = 0; qs < kQSub; ++qs)
#pragma unroll
    for (int c = 0; c < NC; ++c) dq_acc[qs][c] = f32x4{0.f, 0.f, 0.f, 0.f};

  // hoisted staging coordinates + register staging for the K/V tiles
  constexpr int kLVecs = (kKTile * (DP / 8) + kBlockThreads - 1) / kBlockThreads;
  int l_kr[kLVecs], l_dc[kLVecs];
#pragma unroll
  for (int i = 0; i < kLVecs; ++i) {
    const int idx = (int)threadIdx.x + i * kBlockThreads;
    l_kr[i] = idx / (DP / 8);
    l_dc[i] = (idx % (DP / 8)) * 8;
  }
  bf16x8 kst[kLVecs], vst[kLVecs];
  auto issue_kv_loads = [&](int k_base) {
#pragma unroll
    for (int i = 0; i < kLVecs; ++i) {
      bf16x8 kv = {0, 0, 0, 0, 0, 0, 0, 0};
      bf16x8 vv = {0, 0, 0, 0, 0, 0, 0, 0};
      const int k_row = k_base + l_kr[i];
      if (l_kr[i] < kKTile && k_row < T && l_dc[i] < D) {
        kv = *reinterpret_cast<const bf16x8*>(
            &k[qkv_off + (long)k_row * st.qt + l_dc[i]]);
        vv = *reinterpret_cast<const bf16x8*>(
            &v[qkv_off + (long)k_row * st.qt + l_dc[i]]);
      }
      kst[i] = kv;
      vst[i] = vv;
    }
  };
  auto write_kv_tiles = [&]() {
#pragma unroll
    for (int i = 0; i < kLVecs; ++i) {
      if (l_kr[i] >= kKTile) continue;
      *reinterpret_cast<bf16x8*>(&sm.k_tile[l_kr[i]][l_dc[i]]) = kst[i];
      *reinterpret_cast<bf16x8*>(&sm.v_tile[l_kr[i]][l_dc[i]]) = vst[i];
    }
  };

  const int n_ktiles = (T + kKTile - 1) / kKTile;
  issue_kv_loads(0);
  write_kv_tiles();
  __syncthreads();
  for (int kt = 0; kt < n_ktiles; ++kt) {
    const int k_base = kt * kKTile;
    const bool has_next = kt + 1 < n_ktiles;
    // next tile's HBM loads fly under this tile's MFMA phases (T14)
    if (has_next) issue_kv_loads(k_base + kKTile);

#pragma unroll
    for (int qs = 0; qs < kQSub; ++qs) {
      // swapped: S^T = K Q^T, dP^T = V dO^T — C rows = k, cols = q, so
      // each lane owns one q row (q = col) like the forward
      f32x4 st_frag[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
      f32x4 dpt_frag[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int kc = 0; kc < NKC; ++kc) {
          bf16x8 kb = *reinterpret_cast<const bf16x8*>(
              &sm.k_tile[kk * 16 + col][kc * 32 + seg * 8]);
          bf16x8 vb = *reinterpret_cast<const bf16x8*>(
              &sm.v_tile[kk * 16 + col][kc * 32 + seg * 8]);
          st_frag[kk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              kb, q_frag[qs][kc], st_frag[kk], 0, 0, 0);
          dpt_frag[kk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              vb, do_frag[qs][kc], dpt_frag[kk], 0, 0, 0);
        }
      }

      // dS = scale * exp(scale*S - lse) * (dP - Delta), packed to bf16
      // pairs along k (k = k_base + 16*kk + seg*4 + r for q = col)
      unsigned pk[2][2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
          unsigned packed = 0;
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            const int r = 2 * rr + h;
            float ds = 0.f;
            const int k_idx = k_base + 16 * kk + seg * 4 + r;
            if (k_idx < T) {
              const float pv =
                  __expf(st_frag[kk][r] * scale - lse_q[qs]);
              float dp = dpt_frag[kk][r];
              if (DROP) {
                // dP = dPD * D (gradient through the dropout mask)
                const unsigned qg = (unsigned)(bh * T + q_row0 + 16 * qs + col);
                dp = dropout_hash(seed, qg, (unsigned)k_idx) >= drop_thresh
                         ? dp * drop_rscale
                         : 0.f;
              }
              ds = scale * pv * (dp - delta_q[qs]);
            }
            packed |= (unsigned)f32_to_bf16(ds) << (16 * h);
          }
          pk[kk][rr] = packed;
        }
      }
      // redistribute dS to the MFMA A layout (same shuffle pattern as
      // the forward's P)
      const int kk_s = seg >> 1;
      unsigned da_u[4];
#pragma unroll
      for (int j2 = 0; j2 < 4; ++j2) {
        const int srcl = col + 16 * ((seg & 1) * 2 + (j2 >> 1));
        const unsigned lo = (unsigned)__shfl((int)pk[0][j2 & 1], srcl);
        const unsigned hi = (unsigned)__shfl((int)pk[1][j2 & 1], srcl);
        da_u[j2] = kk_s ? hi : lo;
      }
      bf16x8 ds_frag;
      {
        union {
          unsigned u[4];
          bf16x8 v;
        } cvt;
        cvt.u[0] = da_u[0];
        cvt.u[1] = da_u[1];
        cvt.u[2] = da_u[2];
        cvt.u[3] = da_u[3];
        ds_frag = cvt.v;
      }

      // dQ += dS K: B fragments from the row-major K tile via hardware
      // transpose reads (2-deep chunk ring, counted waits)
      const unsigned k_lds_base =
          (unsigned)__builtin_amdgcn_groupstaticsize() +
          (unsigned)offsetof(typename S::SharedDQ, k_tile);
      auto k_tr = [&](int c, int h) -> u32x2 {
        const unsigned addr =
            k_lds_base +
            2u * ((unsigned)((seg * 8 + 4 * h + (col >> 2)) * S::KStride +
                             c * 16 + (col & 3) * 4));
        u32x2 r;
        asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
                     : "=v"(r)
                     : "v"(addr));
        return r;
      };
      u32x2 kf[2][2];
#pragma unroll
      for (int h = 0; h < 2; ++h) kf[0][h] = k_tr(0, h);
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        const int curc = c & 1;
        if (c + 1 < NC) {
#pragma unroll
          for (int h = 0; h < 2; ++h) kf[curc ^ 1][h] = k_tr(c + 1, h);
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(kf[curc][0]), "+v"(kf[curc][1])
                       : [cnt] "i"(2)
                       : "memory");
        } else {
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(kf[curc][0]), "+v"(kf[curc][1])
                       : [cnt] "i"(0)
                       : "memory");
        }
        union {
          u32x2 uu[2];
          bf16x8 v;
        } kc2;
        kc2.uu[0] = kf[curc][0];
        kc2.uu[1] = kf[curc][1];
        dq_acc[qs][c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            ds_frag, kc2.v, dq_acc[qs][c], 0, 0, 0);
      }
    }

    if (has_next) {
      __syncthreads();  // all reads of tile kt done
      write_kv_tiles();
      __syncthreads();  // tile kt+1 staged
    }
  }

#pragma unroll
  for (int qs = 0; qs < kQSub; ++qs) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int q_row = q_row0 + 16 * qs + seg * 4 + r;
      if (q_row >= T) continue;
      const long out_base = qkv_off + (long)q_row * st.qt;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        dq[out_base + c * 16 + col] = (short)f32_to_bf16(dq_acc[qs][c][r]);
      }
    }
  }
}

template <int D, bool DROP = false>
__global__ __launch_bounds__(kBlockThreads, 2) void fmha_bwd_dkv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv, int T, int H,
    QkvStrides st, float scale,
    unsigned drop_thresh = 0, float drop_rscale = 1.f, unsigned seed = 0) {
  using S = FmhaShapes<D>;
  constexpr int DP = S::DP, NKC = S::NKC, NC = S::NC;
  constexpr int kQStream = 32;  // q rows streamed per iteration
  HIP_DYNAMIC_SHARED(char, smem_raw)
  typename S::SharedDKV& sm =
      *reinterpret_cast<typename S::SharedDKV*>(smem_raw);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int seg = lane >> 4;
  const long bh = blockIdx.x;
  const int k_base = blockIdx.y * 64;     // workgroup's 64 k rows
  const int k_row0 = k_base + wave * 16;  // this wave's 16 k rows
  const int b_idx = (int)(bh / H), h_idx = (int)(bh % H);
  const long qkv_off = (long)b_idx * st.qb + (long)h_idx * st.qh;
  const long o_off = (long)b_idx * st.ob + (long)h_idx * st.oh;

  // K and V fragments in registers (A operands of S^T and dP^T)
  bf16x8 k_frag[NKC], v_frag[NKC];
  {
    const int k_row = k_row0 + col;
    const bool valid = k_row < T;
    const long base = qkv_off + (long)k_row * st.qt;
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      const int d0 = kc * 32 + seg * 8;
      if (valid && d0 + 8 <= D) {
        k_frag[kc] = *reinterpret_cast<const bf16x8*>(&k[base + d0]);
        v_frag[kc] = *reinterpret_cast<const bf16x8*>(&v[base + d0]);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const bool ok = valid && d0 + j < D;
          k_frag[kc][j] = ok ? k[base + d0 + j] : (short)0;
          v_frag[kc][j] = ok ? v[base + d0 + j] : (short)0;
        }
      }
    }
  }

  f32x4 dk_acc[NC], dv_acc[NC];
#pragma unroll
  for (int c = 0; c < NC; ++c) {
    dk_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  // hoisted staging coordinates (16B vectors of the q/dO tiles)
  constexpr int kQVecs = (kQStream * (DP / 8) + kBlockThreads - 1) / kBlockThreads;
  int s_qr[kQVecs], s_dc[kQVecs];
#pragma unroll
  for (int i = 0; i < kQVecs; ++i) {
    const int idx = (int)threadIdx.x + i * kBlockThreads;
    s_qr[i] = idx / (DP / 8);
    s_dc[i] = (idx % (DP / 8)) * 8;
  }
  bf16x8 q_st[kQVecs], do_st[kQVecs];
  auto issue_qdo_loads = [&](int q_base) {
#pragma unroll
    for (int i = 0; i < kQVecs; ++i) {
      bf16x8 qv = {0, 0, 0, 0, 0, 0, 0, 0};
      bf16x8 dov = {0, 0, 0, 0, 0, 0, 0, 0};
      const int q_row = q_base + s_qr[i];
      if (s_qr[i] < kQStream && q_row < T && s_dc[i] < D) {
        qv = *reinterpret_cast<const bf16x8*>(
            &q[qkv_off + (long)q_row * st.qt + s_dc[i]]);
        dov = *reinterpret_cast<const bf16x8*>(
            &dout[o_off + (long)q_row * st.ot + s_dc[i]]);
      }
      q_st[i] = qv;
      do_st[i] = dov;
    }
  };
  auto write_qdo_tiles = [&]() {
#pragma unroll
    for (int i = 0; i < kQVecs; ++i) {
      if (s_qr[i] >= kQStream) continue;
      *reinterpret_cast<bf16x8*>(&sm.q_tile[s_qr[i]][s_dc[i]]) = q_st[i];
      *reinterpret_cast<bf16x8*>(&sm.do_tile[s_qr[i]][s_dc[i]]) = do_st[i];
    }
  };

  const int n_qtiles = (T + kQStream - 1) / kQStream;
  issue_qdo_loads(0);
  write_qdo_tiles();
  __syncthreads();
  for (int qt = 0; qt < n_qtiles; ++qt) {
    const int q_base = qt * kQStream;
    const bool has_next = qt + 1 < n_qtiles;
    // next tile's HBM loads fly under this tile's MFMA phases (T14)
    if (has_next) issue_qdo_loads(q_base + kQStream);

    // S^T and dP^T: [16 k][32 q]
    f32x4 st_frag[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
    f32x4 dpt_frag[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll
    for (int qq = 0; qq < 2; ++qq) {
#pragma unroll
      for (int kc = 0; kc < NKC; ++kc) {
        bf16x8 qb = *reinterpret_cast<const bf16x8*>(
            &sm.q_tile[qq * 16 + col][kc * 32 + seg * 8]);
        bf16x8 dob = *reinterpret_cast<const bf16x8*>(
            &sm.do_tile[qq * 16 + col][kc * 32 + seg * 8]);
        st_frag[qq] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            k_frag[kc], qb, st_frag[qq], 0, 0, 0);
        dpt_frag[qq] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            v_frag[kc], dob, dpt_frag[qq], 0, 0, 0);
      }
    }

    // P^T = exp(scale*S^T - lse[q]); dS^T = scale*P^T*(dP^T - Delta[q])
#pragma unroll
    for (int qq = 0; qq < 2; ++qq) {
      const int q_idx = q_base + qq * 16 + col;
      const bool q_ok = q_idx < T;
      const float l = q_ok ? lse[bh * T + q_idx] : 0.f;
      const float dlt = q_ok ? delta[bh * T + q_idx] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pt = 0.f, dst = 0.f;
        if (q_ok) {
          pt = __expf(st_frag[qq][r] * scale - l);
          float dp = dpt_frag[qq][r];
          if (DROP) {
            const unsigned qg = (unsigned)(bh * T + q_idx);
            const unsigned kg = (unsigned)(k_row0 + seg * 4 + r);
            const bool keep = dropout_hash(seed, qg, kg) >= drop_thresh;
            dp = keep ? dp * drop_rscale : 0.f;
            dst = scale * pt * (dp - dlt);
            // dV uses the MASKED P (the forward multiplied PD into V)
            pt = keep ? pt * drop_rscale : 0.f;
          } else {
            dst = scale * pt * (dp - dlt);
          }
        }
        sm.pt_tile[wave][seg * 4 + r][qq * 16 + col] =
            (short)f32_to_bf16(pt);
        sm.dst_tile[wave][seg * 4 + r][qq * 16 + col] =
            (short)f32_to_bf16(dst);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    bf16x8 pt_frag = *reinterpret_cast<const bf16x8*>(
        &sm.pt_tile[wave][col][seg * 8]);
    bf16x8 dst_frag = *reinterpret_cast<const bf16x8*>(
        &sm.dst_tile[wave][col][seg * 8]);
    // dV += P^T dO, dK += dS^T Q: B fragments from the row-major tiles
    // via hardware transpose reads.  TWO sequential passes (dV then dK),
    // each with a single-operand 2-deep ring: the combined dual ring
    // costed 8 more live registers and pushed the kernel to 266 regs /
    // 1 wave/SIMD.
    const unsigned base0 = (unsigned)__builtin_amdgcn_groupstaticsize();
    const unsigned q_lds_base =
        base0 + (unsigned)offsetof(typename S::SharedDKV, q_tile);
    const unsigned do_lds_base =
        base0 + (unsigned)offsetof(typename S::SharedDKV, do_tile);
    auto qdo_tr = [&](unsigned base, int c, int h) -> u32x2 {
      const unsigned addr =
          base + 2u * ((unsigned)((seg * 8 + 4 * h + (col >> 2)) * S::KStride +
                                  c * 16 + (col & 3) * 4));
      u32x2 r;
      asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
                   : "=v"(r)
                   : "v"(addr));
      return r;
    };
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      const unsigned tr_base = pass == 0 ? do_lds_base : q_lds_base;
      const bf16x8 a_frag = pass == 0 ? pt_frag : dst_frag;
      f32x4* accs = pass == 0 ? dv_acc : dk_acc;
      u32x2 bfrag[2][2];
#pragma unroll
      for (int h = 0; h < 2; ++h) bfrag[0][h] = qdo_tr(tr_base, 0, h);
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        const int curc = c & 1;
        if (c + 1 < NC) {
#pragma unroll
          for (int h = 0; h < 2; ++h)
            bfrag[curc ^ 1][h] = qdo_tr(tr_base, c + 1, h);
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(bfrag[curc][0]), "+v"(bfrag[curc][1])
                       : [cnt] "i"(2)
                       : "memory");
        } else {
          asm volatile("s_waitcnt lgkmcnt(%[cnt])"
                       : "+v"(bfrag[curc][0]), "+v"(bfrag[curc][1])
                       : [cnt] "i"(0)
                       : "memory");
        }
        union {
          u32x2 uu[2];
          bf16x8 v;
        } bc2;
        bc2.uu[0] = bfrag[curc][0];
        bc2.uu[1] = bfrag[curc][1];
        accs[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, bc2.v,
                                                          accs[c], 0, 0, 0);
      }
    }
    if (has_next) {
      __syncthreads();  // everyone is done reading tile qt
      write_qdo_tiles();
      __syncthreads();  // tile qt+1 fully staged
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int k_row = k_row0 + seg * 4 + r;
    if (k_row >= T) continue;
    const long out_base = qkv_off + (long)k_row * st.qt;
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      dk[out_base + c * 16 + col] = (short)f32_to_bf16(dk_acc[c][r]);
      dv[out_base + c * 16 + col] = (short)f32_to_bf16(dv_acc[c][r]);
    }
  }
}

// Delta = rowsum(dO * O): one wave per (b,h,t) row; o/dO are strided
// [B,T,H,D]-layout tensors, delta is contiguous [B,H,T]
__global__ __launch_bounds__(64) void fmha_rowdot_kernel(
    const unsigned short* __restrict__ dout,
    const unsigned short* __restrict__ o, float* __restrict__ delta, int T,
    int H, int D, long sob, long soh, long sot) {
  const long row = blockIdx.x;  // linear over [B,H,T]
  const int t = (int)(row % T);
  const int h = (int)((row / T) % H);
  const long b = row / ((long)T * H);
  const long base = b * sob + (long)h * soh + (long)t * sot;
  float acc = 0.f;
  for (int i = threadIdx.x; i < D; i += 64)
    acc += bf16_to_f32(dout[base + i]) * bf16_to_f32(o[base + i]);
  acc = wave_all_sum(acc);
  if (threadIdx.x == 0) delta[row] = acc;
}

// P = exp(scale*S - lse); dS = scale * P * (dP - Delta)
// fused, bf16 io, 8-wide vectorized (rows are T elements, T % 8 == 0 on
// every ViT config; scalar fallback otherwise)
__global__ __launch_bounds__(256) void fmha_dsoftmax_kernel(
    const unsigned short* __restrict__ s_raw,
    const unsigned short* __restrict__ dp, const float* __restrict__ lse,
    const float* __restrict__ delta, unsigned short* __restrict__ p_out,
    unsigned short* __restrict__ ds_out, int T, float scale) {
  const long row = blockIdx.x;  // over B*H*T rows
  const float row_lse = lse[row];
  const float row_delta = delta[row];
  const long base = row * T;
  if ((T & 7) == 0) {
    const int nvec = T / 8;
    const ushort8_t* sv = (const ushort8_t*)(s_raw + base);
    const ushort8_t* dpv = (const ushort8_t*)(dp + base);
    ushort8_t* pv = (ushort8_t*)(p_out + base);
    ushort8_t* dsv = (ushort8_t*)(ds_out + base);
    for (int i = threadIdx.x; i < nvec; i += 256) {
      ushort8_t s8 = sv[i], d8 = dpv[i], p8, ds8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float p = __expf(bf16_to_f32(s8[j]) * scale - row_lse);
        p8[j] = f32_to_bf16(p);
        ds8[j] = f32_to_bf16(scale * p * (bf16_to_f32(d8[j]) - row_delta));
      }
      pv[i] = p8;
      dsv[i] = ds8;
    }
  } else {
    for (int i = threadIdx.x; i < T; i += 256) {
      const float p = __expf(bf16_to_f32(s_raw[base + i]) * scale - row_lse);
      p_out[base + i] = f32_to_bf16(p);
      ds_out[base + i] =
          f32_to_bf16(scale * p * (bf16_to_f32(dp[base + i]) - row_delta));
    }
  }
}


// ds_read_b64_tr_b16 semantics probe: fill LDS with the identity
// pattern lds[i] = i (as raw u16), issue one transpose-read per lane at
// a caller-chosen per-lane base expression, return each lane's 4
// elements.  The GPU test decodes the (lane, elem) -> lds index map so
// kernels can rely on it (no public ISA doc in this environment).
__global__ __launch_bounds__(64) void tr16_probe_kernel(
    unsigned short* __restrict__ out, int mode) {
  HIP_DYNAMIC_SHARED(char, smem_raw)
  unsigned short* lds = reinterpret_cast<unsigned short*>(smem_raw);
  for (int i = threadIdx.x; i < 1024; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  unsigned addr_elems;
  switch (mode) {
    case 0: addr_elems = 0; break;                                // uniform
    case 1: addr_elems = (lane & 15) + (lane >> 4) * 64; break;   // guide map
    case 2: addr_elems = (lane & 15) * 4; break;                  // 4/lane
    default: addr_elems = lane * 4; break;
  }
  // byte address into LDS (dynamic region starts at 0: no static smem)
  unsigned addr = addr_elems * 2 + (unsigned)__builtin_amdgcn_groupstaticsize();
  u32x2 v;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
               : "=v"(v)
               : "v"(addr));
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  out[lane * 4 + 0] = (unsigned short)(v[0] & 0xffff);
  out[lane * 4 + 1] = (unsigned short)(v[0] >> 16);
  out[lane * 4 + 2] = (unsigned short)(v[1] & 0xffff);
  out[lane * 4 + 3] = (unsigned short)(v[1] >> 16);
}

typedef __attribute__((ext_vector_type(16))) float f32x16;

// Layout probe for the WIDE MFMA (mfma_f32_32x32x16_bf16): one wave,
// row-major fp32 A[32][16] / B[16][32], using the assumed gfx9-family
// fragment maps (A: lane l row l&31, k (l>>5)*8+j; C: lane l col l&31,
// row (l>>5)*4 + g*8 + r at acc[g*4+r]).  The GPU test compares against
// torch.matmul to pin the layout before csrc/fgemm.hip relies on it.
__global__ __launch_bounds__(64) void mfma32_probe_kernel(
    const float* __restrict__ a, const float* __restrict__ b,
    float* __restrict__ c) {
  const int l = threadIdx.x;
  union { unsigned short u[8]; bf16x8 v; } av, bv;
  for (int j = 0; j < 8; ++j) {
    av.u[j] = f32_to_bf16(a[(l & 31) * 16 + (l >> 5) * 8 + j]);
    bv.u[j] = f32_to_bf16(b[((l >> 5) * 8 + j) * 32 + (l & 31)]);
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(av.v, bv.v, acc, 0, 0, 0);
  for (int g = 0; g < 4; ++g)
    for (int r = 0; r < 4; ++r)
      c[((l >> 5) * 4 + g * 8 + r) * 32 + (l & 31)] = acc[g * 4 + r];
}

// Layout probe: one wave computes a single 16x16x32 MFMA from row-major
// fp32 A[16][32], B[32][16] using the documented fragment maps; the GPU
// test compares against torch.matmul to pin the layout assumptions.
__global__ __launch_bounds__(64) void mfma_probe_kernel(
    const float* __restrict__ a, const float* __restrict__ b,
    float* __restrict__ c) {
  const int lane = threadIdx.x & 63;
  const int col = lane & 15, seg = lane >> 4;
  bf16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = (short)f32_to_bf16(a[col * 32 + seg * 8 + j]);
    bf[j] = (short)f32_to_bf16(b[(seg * 8 + j) * 16 + col]);
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) c[(seg * 4 + r) * 16 + col] = acc[r];
}

#ifdef VITFSDP_KERNELS_ONLY
// explicit instantiations so -Rpass-analysis reports the real configs
template __global__ void fmha_fwd_kernel<160, false>(
    const short*, const short*, const short*, short*, float*, int, int,
    QkvStrides, float, unsigned, float, unsigned);
template __global__ void fmha_fwd_kernel<160, true>(
    const short*, const short*, const short*, short*, float*, int, int,
    QkvStrides, float, unsigned, float, unsigned);
template __global__ void fmha_fwd_kernel<64, false>(
    const short*, const short*, const short*, short*, float*, int, int,
    QkvStrides, float, unsigned, float, unsigned);
template __global__ void fmha_bwd_dq_kernel<160, false>(
    const short*, const short*, const short*, const short*, const float*,
    const float*, short*, int, int, QkvStrides, float, unsigned, float,
    unsigned);
template __global__ void fmha_bwd_dkv_kernel<160, false>(
    const short*, const short*, const short*, const short*, const float*,
    const float*, short*, short*, int, int, QkvStrides, float, unsigned,
    float, unsigned);
#else
struct FmhaArgs {
  const short *q, *k, *v;
  short* o;        // or dO for backward
  int B, H, T;
  QkvStrides st;
  float scale;
  float p_drop = 0.f;   // attention dropout probability
  unsigned seed = 0;    // dropout mask seed
};

inline unsigned drop_threshold(float p) {
  if (p <= 0.f) return 0u;
  double t = (double)p * 4294967296.0;
  return t >= 4294967295.0 ? 4294967295u : (unsigned)t;
}

template <int D>
void launch_fmha_fwd(const FmhaArgs& a, torch::Tensor& lse) {
  dim3 grid((unsigned)((long)a.B * a.H),
            (a.T + FmhaShapes<D>::QTile - 1) / FmhaShapes<D>::QTile);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (a.p_drop > 0.f) {
    hipLaunchKernelGGL((fmha_fwd_kernel<D, true>), grid, dim3(kBlockThreads),
                       sizeof(typename FmhaShapes<D>::Shared), stream, a.q,
                       a.k, a.v, a.o, lse.data_ptr<float>(), a.T, a.H, a.st,
                       a.scale, drop_threshold(a.p_drop),
                       1.f / (1.f - a.p_drop), a.seed);
  } else {
    hipLaunchKernelGGL((fmha_fwd_kernel<D, false>), grid, dim3(kBlockThreads),
                       sizeof(typename FmhaShapes<D>::Shared), stream, a.q,
                       a.k, a.v, a.o, lse.data_ptr<float>(), a.T, a.H, a.st,
                       a.scale, 0u, 1.f, 0u);
  }
  HIP_CHECK_LAST();
}

template <int D>
void launch_fmha_bwd(const FmhaArgs& a, const torch::Tensor& lse,
                     const torch::Tensor& delta, short* dq, short* dk,
                     short* dv) {
  using S = FmhaShapes<D>;
  auto stream = at::cuda::getCurrentCUDAStream();
  const unsigned BH = (unsigned)((long)a.B * a.H);
  const unsigned thr = drop_threshold(a.p_drop);
  const float rs = a.p_drop > 0.f ? 1.f / (1.f - a.p_drop) : 1.f;
  dim3 grid_dq(BH, (a.T + S::QTile - 1) / S::QTile);
  if (a.p_drop > 0.f) {
    hipLaunchKernelGGL((fmha_bwd_dq_kernel<D, true>), grid_dq,
                       dim3(kBlockThreads), sizeof(typename S::SharedDQ),
                       stream, a.q, a.k, a.v, a.o /*dO*/,
                       lse.data_ptr<float>(), delta.data_ptr<float>(), dq,
                       a.T, a.H, a.st, a.scale, thr, rs, a.seed);
  } else {
    hipLaunchKernelGGL((fmha_bwd_dq_kernel<D, false>), grid_dq,
                       dim3(kBlockThreads), sizeof(typename S::SharedDQ),
                       stream, a.q, a.k, a.v, a.o /*dO*/,
                       lse.data_ptr<float>(), delta.data_ptr<float>(), dq,
                       a.T, a.H, a.st, a.scale, 0u, 1.f, 0u);
  }
  HIP_CHECK_LAST();
  dim3 grid_dkv(BH, (a.T + 63) / 64);
  if (a.p_drop > 0.f) {
    hipLaunchKernelGGL((fmha_bwd_dkv_kernel<D, true>), grid_dkv,
                       dim3(kBlockThreads), sizeof(typename S::SharedDKV),
                       stream, a.q, a.k, a.v, a.o /*dO*/,
                       lse.data_ptr<float>(), delta.data_ptr<float>(), dk, dv,
                       a.T, a.H, a.st, a.scale, thr, rs, a.seed);
  } else {
    hipLaunchKernelGGL((fmha_bwd_dkv_kernel<D, false>), grid_dkv,
                       dim3(kBlockThreads), sizeof(typename S::SharedDKV),
                       stream, a.q, a.k, a.v, a.o /*dO*/,
                       lse.data_ptr<float>(), delta.data_ptr<float>(), dk, dv,
                       a.T, a.H, a.st, a.scale, 0u, 1.f, 0u);
  }
  HIP_CHECK_LAST();
}

#define VITFSDP_FMHA_DISPATCH(D_, expr)                               \
  switch (D_) {                                                       \
    case 16: { constexpr int kD = 16; expr; break; }                  \
    case 32: { constexpr int kD = 32; expr; break; }                  \
    case 48: { constexpr int kD = 48; expr; break; }                  \
    case 64: { constexpr int kD = 64; expr; break; }                  \
    case 80: { constexpr int kD = 80; expr; break; }                  \
    case 96: { constexpr int kD = 96; expr; break; }                  \
    case 112: { constexpr int kD = 112; expr; break; }                \
    case 128: { constexpr int kD = 128; expr; break; }                \
    case 144: { constexpr int kD = 144; expr; break; }                \
    case 160: { constexpr int kD = 160; expr; break; }                \
    case 176: { constexpr int kD = 176; expr; break; }                \
    case 192: { constexpr int kD = 192; expr; break; }                \
    default: TORCH_CHECK(false, "fmha: unsupported head_dim ", D_);   \
  }

#endif  // VITFSDP_KERNELS_ONLY

}  // namespace

#ifndef VITFSDP_KERNELS_ONLY
namespace {

QkvStrides contiguous_strides(int H, int T, int D) {
  QkvStrides st;
  st.qh = (long)T * D;
  st.qb = (long)H * st.qh;
  st.qt = D;
  st.ob = st.qb;
  st.oh = st.qh;
  st.ot = st.qt;
  return st;
}

torch::Tensor run_rowdot(const torch::Tensor& dout, const torch::Tensor& o,
                         int B, int H, int T, int D, const QkvStrides& st) {
  auto delta = torch::empty({B, H, T}, dout.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(fmha_rowdot_kernel, dim3((long)B * H * T), dim3(64), 0,
                     stream, (const unsigned short*)dout.data_ptr(),
                     (const unsigned short*)o.data_ptr(),
                     delta.data_ptr<float>(), T, H, D, st.ob, st.oh, st.ot);
  HIP_CHECK_LAST();
  return delta;
}

}  // namespace

// ---- [B,H,T,D] contiguous API (kernel unit tests, generic use) ----

std::vector<torch::Tensor> fmha_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, double scale,
                                    double p_drop, long seed) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16, "fmha_fwd: bf16 only");
  TORCH_CHECK(q.dim() == 4, "fmha_fwd: [B, H, T, D]");
  const int B = q.size(0), H = q.size(1), T = q.size(2), D = q.size(3);
  TORCH_CHECK(D % 16 == 0 && D <= kMaxD,
              "fmha_fwd: head_dim must be a multiple of 16 and <= ", kMaxD);

  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, T}, q.options().dtype(torch::kFloat32));
  FmhaArgs a{(const short*)q.data_ptr(), (const short*)k.data_ptr(),
             (const short*)v.data_ptr(), (short*)o.data_ptr(),
             B, H, T, contiguous_strides(H, T, D), (float)scale,
             (float)p_drop, (unsigned)seed};
  VITFSDP_FMHA_DISPATCH(D, launch_fmha_fwd<kD>(a, lse));
  return {o, lse};
}

std::vector<torch::Tensor> fmha_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    double scale, double p_drop, long seed) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous());
  const int B = q.size(0), H = q.size(1), T = q.size(2), D = q.size(3);
  const QkvStrides st = contiguous_strides(H, T, D);
  auto delta = run_rowdot(dout, o, B, H, T, D, st);
  auto stream = at::cuda::getCurrentCUDAStream();

  static const bool use_compose = [] {
    const char* e = getenv("VITFSDP_FMHA_BWD");
    return e != nullptr && strcmp(e, "compose") == 0;
  }();
  if (use_compose) {
    // fallback/debug path: batched hipBLASLt GEMMs + fused softmax-grad
    // kernel (measured ~5x slower than the fused kernels at the 10B
    // shape — the tiny per-head batched GEMMs run at 26-50 TF/s)
    auto s_raw = at::matmul(q, k.transpose(-2, -1));
    auto dp = at::matmul(dout, v.transpose(-2, -1));
    auto p = torch::empty_like(s_raw);
    auto ds = torch::empty_like(s_raw);
    hipLaunchKernelGGL(fmha_dsoftmax_kernel, dim3((long)B * H * T), dim3(256),
                       0, stream, (const unsigned short*)s_raw.data_ptr(),
                       (const unsigned short*)dp.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)p.data_ptr(),
                       (unsigned short*)ds.data_ptr(), T, (float)scale);
    HIP_CHECK_LAST();
    auto dv = at::matmul(p.transpose(-2, -1), dout);
    auto dq = at::matmul(ds, k);
    auto dk = at::matmul(ds.transpose(-2, -1), q);
    return {dq, dk, dv};
  }

  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  FmhaArgs a{(const short*)q.data_ptr(), (const short*)k.data_ptr(),
             (const short*)v.data_ptr(), (short*)dout.data_ptr(),
             B, H, T, st, (float)scale, (float)p_drop, (unsigned)seed};
  VITFSDP_FMHA_DISPATCH(
      D, launch_fmha_bwd<kD>(a, lse, delta, (short*)dq.data_ptr(),
                             (short*)dk.data_ptr(), (short*)dv.data_ptr()));
  return {dq, dk, dv};
}

// ---- fused-qkv API: zero-copy path used by the Attention module ----
// qkv: [B, T, 3, H, D] contiguous (the fused projection output reshaped);
// o comes back [B, T, H, D] (== [B, T, E]); dO arrives the same way.

std::vector<torch::Tensor> fmha_fwd_qkv(torch::Tensor qkv, long num_heads,
                                        double scale, double p_drop,
                                        long seed) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous() && qkv.dim() == 5 &&
                  qkv.size(2) == 3 && qkv.size(3) == num_heads,
              "fmha_fwd_qkv: expected contiguous [B, T, 3, H, D]");
  TORCH_CHECK(qkv.scalar_type() == torch::kBFloat16, "fmha: bf16 only");
  const int B = qkv.size(0), T = qkv.size(1), H = qkv.size(3),
            D = qkv.size(4);
  TORCH_CHECK(D % 16 == 0 && D <= kMaxD,
              "fmha: head_dim must be a multiple of 16 and <= ", kMaxD);
  QkvStrides st;
  st.qt = 3L * H * D;
  st.qb = (long)T * st.qt;
  st.qh = D;
  st.ot = (long)H * D;
  st.ob = (long)T * st.ot;
  st.oh = D;

  auto o = torch::empty({B, T, (long)H * D}, qkv.options());
  auto lse = torch::empty({B, H, T}, qkv.options().dtype(torch::kFloat32));
  const short* base = (const short*)qkv.data_ptr();
  FmhaArgs a{base, base + (long)H * D, base + 2L * H * D,
             (short*)o.data_ptr(), B, H, T, st, (float)scale,
             (float)p_drop, (unsigned)seed};
  VITFSDP_FMHA_DISPATCH(D, launch_fmha_fwd<kD>(a, lse));
  return {o, lse};
}

torch::Tensor fmha_bwd_qkv(torch::Tensor dout, torch::Tensor qkv,
                           torch::Tensor o, torch::Tensor lse,
                           long num_heads, double scale, double p_drop,
                           long seed) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous() && qkv.is_contiguous());
  const int B = qkv.size(0), T = qkv.size(1), H = qkv.size(3),
            D = qkv.size(4);
  QkvStrides st;
  st.qt = 3L * H * D;
  st.qb = (long)T * st.qt;
  st.qh = D;
  st.ot = (long)H * D;
  st.ob = (long)T * st.ot;
  st.oh = D;
  auto delta = run_rowdot(dout, o, B, H, T, D, st);

  auto dqkv = torch::empty_like(qkv);
  const short* base = (const short*)qkv.data_ptr();
  short* dbase = (short*)dqkv.data_ptr();
  FmhaArgs a{base, base + (long)H * D, base + 2L * H * D,
             (short*)dout.data_ptr(), B, H, T, st, (float)scale,
             (float)p_drop, (unsigned)seed};
  VITFSDP_FMHA_DISPATCH(
      D, launch_fmha_bwd<kD>(a, lse, delta, dbase, dbase + (long)H * D,
                             dbase + 2L * H * D));
  return dqkv;
}


torch::Tensor tr16_probe(long mode) {
  auto out = torch::zeros({64, 4}, torch::TensorOptions()
                                       .dtype(torch::kInt16)
                                       .device(torch::kCUDA));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 2048, stream,
                     (unsigned short*)out.data_ptr(), (int)mode);
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor mfma32_probe(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && a.dim() == 2 &&
              a.size(0) == 32 && a.size(1) == 16);
  TORCH_CHECK(b.is_cuda() && b.is_contiguous() && b.dim() == 2 &&
              b.size(0) == 16 && b.size(1) == 32);
  TORCH_CHECK(a.scalar_type() == torch::kFloat32 &&
              b.scalar_type() == torch::kFloat32);
  auto c = torch::zeros({32, 32}, a.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(mfma32_probe_kernel, dim3(1), dim3(64), 0, stream,
                     a.data_ptr<float>(), b.data_ptr<float>(),
                     c.data_ptr<float>());
  HIP_CHECK_LAST();
  return c;
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(b.is_cuda() && b.sizes() == torch::IntArrayRef({32, 16}));
  auto a32 = a.to(torch::kFloat32).contiguous();
  auto b32 = b.to(torch::kFloat32).contiguous();
  auto c = torch::empty({16, 16}, a32.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     a32.data_ptr<float>(), b32.data_ptr<float>(),
                     c.data_ptr<float>());
  HIP_CHECK_LAST();
  return c;
}
#endif  // VITFSDP_KERNELS_ONLY
