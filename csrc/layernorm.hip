// LayerNorm forward/backward for CDNA4 (kernel K2).
//
// Replaces the torch LayerNorm in every ViT block (reference: timm Block
// norm1/norm2 and the final norm, run_vit_training.py:151).  Memory-bound:
// rows of D bf16 elements, fp32 statistics, 16-byte vectorized loads
// (guide G13: scalar bf16 loads cost ~2x).  One 256-thread block per row;
// the second pass re-reads x through L2 (a row is ~10 KB at D=5120, far
// inside the 4 MB per-XCD L2), which keeps register pressure flat across
// any D instead of caching the row in registers.

#ifndef VITFSDP_KERNELS_ONLY
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>
#endif

#include "common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kWaves = kBlock / WAVE_SIZE;

__global__ void ln_fwd_kernel(const ushort8_t* __restrict__ x,
                              const ushort8_t* __restrict__ w,
                              const ushort8_t* __restrict__ b,
                              ushort8_t* __restrict__ y,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out, int nvec,
                              float inv_d, float eps) {
  __shared__ float scratch[kWaves];
  const long row = blockIdx.x;
  const ushort8_t* xr = x + row * nvec;
  ushort8_t* yr = y + row * nvec;

  float s = 0.f, ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    ushort8_t v = xr[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(v[j]);
      s += f;
      ss += f * f;
    }
  }
  s = block_sum<kWaves>(s, scratch);
  ss = block_sum<kWaves>(ss, scratch);
  const float mean = s * inv_d;
  const float var = fmaxf(ss * inv_d - mean * mean, 0.f);
  const float rstd = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }

  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    ushort8_t xv = xr[i], wv = w[i], bv = b[i], out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xh = (bf16_to_f32(xv[j]) - mean) * rstd;
      out[j] = f32_to_bf16(xh * bf16_to_f32(wv[j]) + bf16_to_f32(bv[j]));
    }
    yr[i] = out;
  }
}

// dx = rstd * (g - mean(g) - xhat * mean(g*xhat)), g = dy * w
__global__ void ln_bwd_dx_kernel(const ushort8_t* __restrict__ dy,
                                 const ushort8_t* __restrict__ x,
                                 const ushort8_t* __restrict__ w,
                                 const float* __restrict__ mean_in,
                                 const float* __restrict__ rstd_in,
                                 ushort8_t* __restrict__ dx, int nvec,
                                 float inv_d) {
  __shared__ float scratch[kWaves];
  const long row = blockIdx.x;
  const ushort8_t* dyr = dy + row * nvec;
  const ushort8_t* xr = x + row * nvec;
  ushort8_t* dxr = dx + row * nvec;
  const float mean = mean_in[row];
  const float rstd = rstd_in[row];

  float sg = 0.f, sgx = 0.f;
  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    ushort8_t dv = dyr[i], xv = xr[i], wv = w[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf16_to_f32(dv[j]) * bf16_to_f32(wv[j]);
      float xh = (bf16_to_f32(xv[j]) - mean) * rstd;
      sg += g;
      sgx += g * xh;
    }
  }
  sg = block_sum<kWaves>(sg, scratch) * inv_d;
  sgx = block_sum<kWaves>(sgx, scratch) * inv_d;

  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    ushort8_t dv = dyr[i], xv = xr[i], wv = w[i], out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf16_to_f32(dv[j]) * bf16_to_f32(wv[j]);
      float xh = (bf16_to_f32(xv[j]) - mean) * rstd;
      out[j] = f32_to_bf16(rstd * (g - sg - xh * sgx));
    }
    dxr[i] = out;
  }
}

// Column reductions for dgamma/dbeta: grid (D/256, row_chunks); each
// thread owns one column within its row chunk (coalesced across the
// 256 consecutive columns of the block), partials in fp32.
__global__ void ln_bwd_dwdb_partial_kernel(
    const unsigned short* __restrict__ dy, const unsigned short* __restrict__ x,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    float* __restrict__ dw_part, float* __restrict__ db_part, int d,
    long n_rows, int rows_per_chunk) {
  const int col = blockIdx.x * kBlock + threadIdx.x;
  if (col >= d) return;
  const long row_begin = (long)blockIdx.y * rows_per_chunk;
  const long row_end = min(row_begin + rows_per_chunk, n_rows);
  float dw = 0.f, db = 0.f;
  for (long r = row_begin; r < row_end; ++r) {
    float dyv = bf16_to_f32(dy[r * d + col]);
    float xh = (bf16_to_f32(x[r * d + col]) - mean_in[r]) * rstd_in[r];
    dw += dyv * xh;
    db += dyv;
  }
  dw_part[(long)blockIdx.y * d + col] = dw;
  db_part[(long)blockIdx.y * d + col] = db;
}

__global__ void ln_bwd_dwdb_final_kernel(const float* __restrict__ dw_part,
                                         const float* __restrict__ db_part,
                                         unsigned short* __restrict__ dw,
                                         unsigned short* __restrict__ db,
                                         int d, int n_chunks) {
  const int col = blockIdx.x * kBlock + threadIdx.x;
  if (col >= d) return;
  float sw = 0.f, sb = 0.f;
  for (int c = 0; c < n_chunks; ++c) {
    sw += dw_part[(long)c * d + col];
    sb += db_part[(long)c * d + col];
  }
  dw[col] = f32_to_bf16(sw);
  db[col] = f32_to_bf16(sb);
}


// Fused residual-add + LayerNorm (SURVEY.md K2 fused epilogue): the
// pre-LN block computes norm2(x + attn_out) — fusing the add into the
// LN's statistics pass removes one full read+write of the [N, D] tensor
// per call, and the backward folds the residual gradient add into the
// dx pass (no separate elementwise add kernels).
__global__ void ln_add_fwd_kernel(const ushort8_t* __restrict__ x,
                                  const ushort8_t* __restrict__ res,
                                  const ushort8_t* __restrict__ w,
                                  const ushort8_t* __restrict__ b,
                                  ushort8_t* __restrict__ sum_out,
                                  ushort8_t* __restrict__ y,
                                  float* __restrict__ mean_out,
                                  float* __restrict__ rstd_out, int nvec,
                                  float inv_d, float eps) {
  __shared__ float scratch[kWaves];
  const long row = blockIdx.x;
  const ushort8_t* xr = x + row * nvec;
  const ushort8_t* rr = res + row * nvec;
  ushort8_t* sr = sum_out + row * nvec;
  ushort8_t* yr = y + row * nvec;

  float s = 0.f, ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    ushort8_t xv = xr[i], rv = rr[i], sv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(xv[j]) + bf16_to_f32(rv[j]);
      sv[j] = f32_to_bf16(f);
      // statistics on the bf16-rounded sum, so pass 2 normalizes the
      // exact values it re-reads
      f = bf16_to_f32(sv[j]);
      s += f;
      ss += f * f;
    }
    sr[i] = sv;
  }
  s = block_sum<kWaves>(s, scratch);
  ss = block_sum<kWaves>(ss, scratch);
  const float mean = s * inv_d;
  const float var = fmaxf(ss * inv_d - mean * mean, 0.f);
  const float rstd = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }

  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    ushort8_t sv = sr[i], wv = w[i], bv = b[i], out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xh = (bf16_to_f32(sv[j]) - mean) * rstd;
      out[j] = f32_to_bf16(xh * bf16_to_f32(wv[j]) + bf16_to_f32(bv[j]));
    }
    yr[i] = out;
  }
}

// dx = rstd*(g - mean(g) - xhat*mean(g*xhat)) + dsum; written to TWO
// buffers (the gradients of x and res are identical but the autograd
// engine must receive distinct tensors it may mutate independently)
__global__ void ln_add_bwd_dx_kernel(const ushort8_t* __restrict__ dy,
                                     const ushort8_t* __restrict__ dsum,
                                     const ushort8_t* __restrict__ s_in,
                                     const ushort8_t* __restrict__ w,
                                     const float* __restrict__ mean_in,
                                     const float* __restrict__ rstd_in,
                                     ushort8_t* __restrict__ dx1,
                                     ushort8_t* __restrict__ dx2, int nvec,
                                     float inv_d) {
  __shared__ float scratch[kWaves];
  const long row = blockIdx.x;
  const ushort8_t* dyr = dy + row * nvec;
  const ushort8_t* dsr = dsum + row * nvec;
  const ushort8_t* sr = s_in + row * nvec;
  ushort8_t* d1 = dx1 + row * nvec;
  ushort8_t* d2 = dx2 + row * nvec;
  const float mean = mean_in[row];
  const float rstd = rstd_in[row];

  float sg = 0.f, sgx = 0.f;
  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    ushort8_t dv = dyr[i], sv = sr[i], wv = w[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf16_to_f32(dv[j]) * bf16_to_f32(wv[j]);
      float xh = (bf16_to_f32(sv[j]) - mean) * rstd;
      sg += g;
      sgx += g * xh;
    }
  }
  sg = block_sum<kWaves>(sg, scratch) * inv_d;
  sgx = block_sum<kWaves>(sgx, scratch) * inv_d;

  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    ushort8_t dv = dyr[i], sv = sr[i], wv = w[i], av = dsr[i], out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf16_to_f32(dv[j]) * bf16_to_f32(wv[j]);
      float xh = (bf16_to_f32(sv[j]) - mean) * rstd;
      out[j] = f32_to_bf16(rstd * (g - sg - xh * sgx) + bf16_to_f32(av[j]));
    }
    d1[i] = out;
    d2[i] = out;
  }
}

}  // namespace

#ifndef VITFSDP_KERNELS_ONLY
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "layernorm_fwd: bf16 only");
  const int d = x.size(-1);
  TORCH_CHECK(d % 8 == 0, "layernorm_fwd: D must be a multiple of 8");
  const long n = x.numel() / d;
  auto y = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({n}, opts);
  auto rstd = torch::empty({n}, opts);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(ln_fwd_kernel, dim3((unsigned)n), dim3(kBlock), 0, stream,
                     (const ushort8_t*)x.data_ptr(),
                     (const ushort8_t*)w.data_ptr(),
                     (const ushort8_t*)b.data_ptr(), (ushort8_t*)y.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), d / 8,
                     1.f / d, (float)eps);
  HIP_CHECK_LAST();
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  const int d = x.size(-1);
  const long n = x.numel() / d;
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto db = torch::empty_like(w);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(ln_bwd_dx_kernel, dim3((unsigned)n), dim3(kBlock), 0,
                     stream, (const ushort8_t*)dy.data_ptr(),
                     (const ushort8_t*)x.data_ptr(),
                     (const ushort8_t*)w.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), (ushort8_t*)dx.data_ptr(), d / 8,
                     1.f / d);
  HIP_CHECK_LAST();

  // pick row chunking so the partial grid fills the chip: target ~2048
  // blocks total (256 CUs x 8, guide G11)
  const int grid_x = (d + kBlock - 1) / kBlock;
  int n_chunks = std::max(1, 2048 / std::max(grid_x, 1));
  int rows_per_chunk = (int)((n + n_chunks - 1) / n_chunks);
  rows_per_chunk = std::max(rows_per_chunk, 16);
  n_chunks = (int)((n + rows_per_chunk - 1) / rows_per_chunk);
  auto opts = x.options().dtype(torch::kFloat32);
  auto dw_part = torch::empty({n_chunks, d}, opts);
  auto db_part = torch::empty({n_chunks, d}, opts);
  dim3 grid((d + kBlock - 1) / kBlock, n_chunks);
  hipLaunchKernelGGL(ln_bwd_dwdb_partial_kernel, grid, dim3(kBlock), 0, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     dw_part.data_ptr<float>(), db_part.data_ptr<float>(), d, n,
                     rows_per_chunk);
  HIP_CHECK_LAST();
  hipLaunchKernelGGL(ln_bwd_dwdb_final_kernel, dim3((d + kBlock - 1) / kBlock),
                     dim3(kBlock), 0, stream, dw_part.data_ptr<float>(),
                     db_part.data_ptr<float>(),
                     (unsigned short*)dw.data_ptr(),
                     (unsigned short*)db.data_ptr(), d, n_chunks);
  HIP_CHECK_LAST();
  return {dx, dw, db};
}


std::vector<torch::Tensor> layernorm_add_fwd(torch::Tensor x,
                                             torch::Tensor res,
                                             torch::Tensor w, torch::Tensor b,
                                             double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && res.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "bf16 only");
  const int d = x.size(-1);
  TORCH_CHECK(d % 8 == 0, "layernorm_add_fwd: D must be a multiple of 8");
  const long n = x.numel() / d;
  auto sum = torch::empty_like(x);
  auto y = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({n}, opts);
  auto rstd = torch::empty({n}, opts);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(ln_add_fwd_kernel, dim3((unsigned)n), dim3(kBlock), 0,
                     stream, (const ushort8_t*)x.data_ptr(),
                     (const ushort8_t*)res.data_ptr(),
                     (const ushort8_t*)w.data_ptr(),
                     (const ushort8_t*)b.data_ptr(),
                     (ushort8_t*)sum.data_ptr(), (ushort8_t*)y.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), d / 8,
                     1.f / d, (float)eps);
  HIP_CHECK_LAST();
  return {sum, y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_add_bwd(torch::Tensor dy,
                                             torch::Tensor dsum,
                                             torch::Tensor sum,
                                             torch::Tensor w,
                                             torch::Tensor mean,
                                             torch::Tensor rstd) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dsum.is_contiguous() &&
              sum.is_contiguous());
  const int d = sum.size(-1);
  const long n = sum.numel() / d;
  auto dx1 = torch::empty_like(sum);
  auto dx2 = torch::empty_like(sum);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(ln_add_bwd_dx_kernel, dim3((unsigned)n), dim3(kBlock), 0,
                     stream, (const ushort8_t*)dy.data_ptr(),
                     (const ushort8_t*)dsum.data_ptr(),
                     (const ushort8_t*)sum.data_ptr(),
                     (const ushort8_t*)w.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), (ushort8_t*)dx1.data_ptr(),
                     (ushort8_t*)dx2.data_ptr(), d / 8, 1.f / d);
  HIP_CHECK_LAST();

  // dgamma/dbeta: same two-stage column reduction as plain LN, with the
  // summed input playing the role of x
  auto dw = torch::empty_like(w);
  auto db = torch::empty_like(w);
  const int grid_x = (d + kBlock - 1) / kBlock;
  int n_chunks = std::max(1, 2048 / std::max(grid_x, 1));
  int rows_per_chunk = (int)((n + n_chunks - 1) / n_chunks);
  rows_per_chunk = std::max(rows_per_chunk, 16);
  n_chunks = (int)((n + rows_per_chunk - 1) / rows_per_chunk);
  auto opts = sum.options().dtype(torch::kFloat32);
  auto dw_part = torch::empty({n_chunks, d}, opts);
  auto db_part = torch::empty({n_chunks, d}, opts);
  dim3 grid(grid_x, n_chunks);
  hipLaunchKernelGGL(ln_bwd_dwdb_partial_kernel, grid, dim3(kBlock), 0, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)sum.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     dw_part.data_ptr<float>(), db_part.data_ptr<float>(), d,
                     n, rows_per_chunk);
  HIP_CHECK_LAST();
  hipLaunchKernelGGL(ln_bwd_dwdb_final_kernel, dim3(grid_x), dim3(kBlock), 0,
                     stream, dw_part.data_ptr<float>(),
                     db_part.data_ptr<float>(), (unsigned short*)dw.data_ptr(),
                     (unsigned short*)db.data_ptr(), d, n_chunks);
  HIP_CHECK_LAST();
  return {dx1, dx2, dw, db};
}

#endif  // VITFSDP_KERNELS_ONLY
