// Fused softmax cross-entropy (kernel K9): log-softmax +
// NLL in one pass over bf16 logits with fp32 accumulation, mean
// reduction (reference: torch.nn.CrossEntropyLoss at
// run_vit_training.py:229,262).  Shapes are small ([batch, 1000]); one
// 256-thread block per row, loss accumulated with one fp32 atomic.

#ifndef VITFSDP_KERNELS_ONLY
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>
#endif

#include "common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kWaves = kBlock / WAVE_SIZE;

__global__ void ce_fwd_kernel(const unsigned short* __restrict__ logits,
                              const long* __restrict__ target,
                              float* __restrict__ loss_out,
                              float* __restrict__ lse_out, int c,
                              float inv_n) {
  __shared__ float scratch[kWaves];
  const long row = blockIdx.x;
  const unsigned short* lr = logits + row * c;

  float m = -INFINITY;
  for (int i = threadIdx.x; i < c; i += kBlock)
    m = fmaxf(m, bf16_to_f32(lr[i]));
  m = block_max<kWaves>(m, scratch);

  float s = 0.f;
  for (int i = threadIdx.x; i < c; i += kBlock)
    s += __expf(bf16_to_f32(lr[i]) - m);
  s = block_sum<kWaves>(s, scratch);

  const float lse = m + __logf(s);
  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    const float tl = bf16_to_f32(lr[target[row]]);
    atomicAdd(loss_out, (lse - tl) * inv_n);
  }
}

__global__ void ce_bwd_kernel(const unsigned short* __restrict__ logits,
                              const long* __restrict__ target,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,
                              unsigned short* __restrict__ dlogits, int c,
                              float inv_n) {
  const long row = blockIdx.x;
  const unsigned short* lr = logits + row * c;
  unsigned short* dr = dlogits + row * c;
  const float scale = (*dloss) * inv_n;
  const float row_lse = lse[row];
  const long tgt = target[row];
  for (int i = threadIdx.x; i < c; i += kBlock) {
    float p = __expf(bf16_to_f32(lr[i]) - row_lse);
    float d = (p - (i == tgt ? 1.f : 0.f)) * scale;
    dr[i] = f32_to_bf16(d);
  }
}

}  // namespace

#ifndef VITFSDP_KERNELS_ONLY
std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits,
                                             torch::Tensor target) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 2);
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16,
              "cross_entropy_fwd: bf16 logits only");
  TORCH_CHECK(target.scalar_type() == torch::kLong);
  const long n = logits.size(0);
  const int c = (int)logits.size(1);
  auto opts = logits.options().dtype(torch::kFloat32);
  auto loss = torch::zeros({}, opts);
  auto lse = torch::empty({n}, opts);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(ce_fwd_kernel, dim3((unsigned)n), dim3(kBlock), 0, stream,
                     (const unsigned short*)logits.data_ptr(),
                     target.data_ptr<long>(), loss.data_ptr<float>(),
                     lse.data_ptr<float>(), c, 1.f / n);
  HIP_CHECK_LAST();
  return {loss, lse};
}

torch::Tensor cross_entropy_bwd(torch::Tensor dloss, torch::Tensor logits,
                                torch::Tensor target, torch::Tensor lse) {
  const long n = logits.size(0);
  const int c = (int)logits.size(1);
  auto dlogits = torch::empty_like(logits);
  auto dloss_f = dloss.to(torch::kFloat32);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(ce_bwd_kernel, dim3((unsigned)n), dim3(kBlock), 0, stream,
                     (const unsigned short*)logits.data_ptr(),
                     target.data_ptr<long>(), lse.data_ptr<float>(),
                     dloss_f.data_ptr<float>(),
                     (unsigned short*)dlogits.data_ptr(), c, 1.f / n);
  HIP_CHECK_LAST();
  return dlogits;
}

#endif  // VITFSDP_KERNELS_ONLY
