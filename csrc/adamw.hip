// Fused multi-tensor kernels over the FSDP fp32 master shards:
//   * AdamW step (kernel K8) — decoupled weight decay,
//     identical math to torch.optim.AdamW (reference run_vit_training.py:237)
//   * grad sq-norm + scale (kernel K11) backing clip_grad_norm_
//     (reference run_vit_training.py:270)
//
// The FSDP engine gives one flat shard per wrapped unit (~33 large fp32
// tensors for ViT-10B), so the multi-tensor scheme is simple: the tensor
// table rides in the kernarg segment by value and a fixed grid strides
// over each tensor in turn with float4 (16 B) accesses.  Purely
// memory-bound; one launch per <=32 tensors.

#ifndef VITFSDP_KERNELS_ONLY
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>
#endif

#include "common.h"

namespace {

constexpr int kMaxTensors = 32;
constexpr int kBlock = 256;
constexpr int kGrid = 2048;  // 256 CUs x 8 blocks (guide G11)

struct AdamWTable {
  float* p[kMaxTensors];
  const float* g[kMaxTensors];
  float* m[kMaxTensors];
  float* v[kMaxTensors];
  unsigned short* mir[kMaxTensors];  // optional bf16 mirror of p
  long n[kMaxTensors];
  int count;
};

struct TensorTable {
  float* t[kMaxTensors];
  long n[kMaxTensors];
  int count;
};

__global__ void fused_adamw_kernel(AdamWTable tab, float lr, float beta1,
                                   float beta2, float eps, float wd,
                                   float bias_c1, float bias_c2,
                                   const float* __restrict__ grad_scale) {
  // deferred gradient clipping: the clip coefficient (a device scalar
  // from clip_grad_norm_(defer_scale=True)) scales g on the fly, which
  // replaces a separate full read+write pass over every gradient shard
  const float gs = grad_scale ? *grad_scale : 1.f;
  const float decay = 1.f - lr * wd;
  const float step_size = lr / bias_c1;
  const float inv_sqrt_c2 = rsqrtf(bias_c2);
  for (int ti = 0; ti < tab.count; ++ti) {
    float4* p = (float4*)tab.p[ti];
    const float4* g = (const float4*)tab.g[ti];
    float4* m = (float4*)tab.m[ti];
    float4* v = (float4*)tab.v[ti];
    const long n4 = tab.n[ti] / 4;
    for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n4;
         i += (long)gridDim.x * kBlock) {
      float4 pv = p[i], gv = g[i], mv = m[i], vv = v[i];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float pj = (&pv.x)[j], gj = (&gv.x)[j] * gs;
        float mj = beta1 * (&mv.x)[j] + (1.f - beta1) * gj;
        float vj = beta2 * (&vv.x)[j] + (1.f - beta2) * gj * gj;
        pj *= decay;
        pj -= step_size * mj / (sqrtf(vj) * inv_sqrt_c2 + eps);
        (&pv.x)[j] = pj;
        (&mv.x)[j] = mj;
        (&vv.x)[j] = vj;
      }
      p[i] = pv;
      m[i] = mv;
      v[i] = vv;
      if (tab.mir[ti]) {
        // keep the bf16 comm mirror in sync so the FSDP gather path
        // never re-casts the fp32 master
        struct Bf16x4 {
          unsigned short d[4];
        } mb;
#pragma unroll
        for (int j = 0; j < 4; ++j) mb.d[j] = f32_to_bf16((&pv.x)[j]);
        ((Bf16x4*)tab.mir[ti])[i] = mb;
      }
    }
    // scalar tail (shards are padded to world_size, not necessarily to 4)
    const long tail = tab.n[ti] & 3;
    if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
      const long i = tab.n[ti] - tail + threadIdx.x;
      float pj = tab.p[ti][i], gj = tab.g[ti][i] * gs;
      float mj = beta1 * tab.m[ti][i] + (1.f - beta1) * gj;
      float vj = beta2 * tab.v[ti][i] + (1.f - beta2) * gj * gj;
      pj = pj * decay - step_size * mj / (sqrtf(vj) * inv_sqrt_c2 + eps);
      tab.p[ti][i] = pj;
      tab.m[ti][i] = mj;
      tab.v[ti][i] = vj;
      if (tab.mir[ti]) tab.mir[ti][i] = f32_to_bf16(pj);
    }
  }
}

__global__ void mt_sqnorm_kernel(TensorTable tab, float* __restrict__ out) {
  __shared__ float scratch[kBlock / WAVE_SIZE];
  float acc = 0.f;
  for (int ti = 0; ti < tab.count; ++ti) {
    const float4* t = (const float4*)tab.t[ti];
    const long n4 = tab.n[ti] / 4;
    for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n4;
         i += (long)gridDim.x * kBlock) {
      float4 v = t[i];
      acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
    const long tail = tab.n[ti] & 3;
    if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
      float v = tab.t[ti][tab.n[ti] - tail + threadIdx.x];
      acc += v * v;
    }
  }
  acc = block_sum<kBlock / WAVE_SIZE>(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

__global__ void mt_scale_kernel(TensorTable tab,
                                const float* __restrict__ factor_ptr,
                                float factor_imm, int use_ptr) {
  const float f = use_ptr ? *factor_ptr : factor_imm;
  for (int ti = 0; ti < tab.count; ++ti) {
    float4* t = (float4*)tab.t[ti];
    const long n4 = tab.n[ti] / 4;
    for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n4;
         i += (long)gridDim.x * kBlock) {
      float4 v = t[i];
      v.x *= f;
      v.y *= f;
      v.z *= f;
      v.w *= f;
      t[i] = v;
    }
    const long tail = tab.n[ti] & 3;
    if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
      tab.t[ti][tab.n[ti] - tail + threadIdx.x] *= f;
    }
  }
}

int grid_for(long total_elems) {
  long blocks = (total_elems / 4 + kBlock - 1) / kBlock;
  return (int)std::max(1L, std::min((long)kGrid, blocks));
}

}  // namespace

#ifndef VITFSDP_KERNELS_ONLY
void fused_adamw(std::vector<torch::Tensor> params,
                 std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs,
                 std::vector<torch::Tensor> exp_avg_sqs,
                 std::vector<c10::optional<torch::Tensor>> mirrors,
                 double lr,
                 double beta1, double beta2, double eps, double weight_decay,
                 double bias_c1, double bias_c2,
                 c10::optional<torch::Tensor> grad_scale) {
  TORCH_CHECK(mirrors.empty() || mirrors.size() == params.size(),
              "mirrors must be empty or match params");
  const float* gs_ptr = nullptr;
  if (grad_scale.has_value()) {
    TORCH_CHECK(grad_scale->is_cuda() &&
                    grad_scale->scalar_type() == torch::kFloat32,
                "grad_scale must be a CUDA fp32 scalar");
    gs_ptr = grad_scale->data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  const int n = (int)params.size();
  for (int base = 0; base < n; base += kMaxTensors) {
    AdamWTable tab;
    tab.count = std::min(kMaxTensors, n - base);
    long total = 0;
    for (int i = 0; i < tab.count; ++i) {
      auto& p = params[base + i];
      TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kFloat32 &&
                      p.is_contiguous(),
                  "fused_adamw expects contiguous fp32 CUDA tensors");
      tab.p[i] = p.data_ptr<float>();
      tab.g[i] = grads[base + i].data_ptr<float>();
      tab.m[i] = exp_avgs[base + i].data_ptr<float>();
      tab.v[i] = exp_avg_sqs[base + i].data_ptr<float>();
      tab.mir[i] = nullptr;
      if (!mirrors.empty() && mirrors[base + i].has_value()) {
        auto& mt = *mirrors[base + i];
        TORCH_CHECK(mt.scalar_type() == torch::kBFloat16 &&
                        mt.numel() == p.numel() && mt.is_contiguous(),
                    "mirror must be a contiguous bf16 tensor of p's size");
        tab.mir[i] = (unsigned short*)mt.data_ptr();
      }
      tab.n[i] = p.numel();
      total += tab.n[i];
    }
    hipLaunchKernelGGL(fused_adamw_kernel, dim3(grid_for(total)), dim3(kBlock),
                       0, stream, tab, (float)lr, (float)beta1, (float)beta2,
                       (float)eps, (float)weight_decay, (float)bias_c1,
                       (float)bias_c2, gs_ptr);
    HIP_CHECK_LAST();
  }
}

torch::Tensor multi_tensor_sqnorm(std::vector<torch::Tensor> tensors) {
  auto stream = at::cuda::getCurrentCUDAStream();
  auto out = torch::zeros({}, tensors[0].options().dtype(torch::kFloat32));
  const int n = (int)tensors.size();
  for (int base = 0; base < n; base += kMaxTensors) {
    TensorTable tab;
    tab.count = std::min(kMaxTensors, n - base);
    long total = 0;
    for (int i = 0; i < tab.count; ++i) {
      auto& t = tensors[base + i];
      TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                      t.is_contiguous(),
                  "multi_tensor_sqnorm expects contiguous fp32 CUDA tensors");
      tab.t[i] = t.data_ptr<float>();
      tab.n[i] = t.numel();
      total += tab.n[i];
    }
    hipLaunchKernelGGL(mt_sqnorm_kernel, dim3(grid_for(total)), dim3(kBlock), 0,
                       stream, tab, out.data_ptr<float>());
    HIP_CHECK_LAST();
  }
  return out;
}

static void mt_scale_impl(std::vector<torch::Tensor>& tensors,
                          const float* factor_ptr, float factor_imm) {
  auto stream = at::cuda::getCurrentCUDAStream();
  const int n = (int)tensors.size();
  for (int base = 0; base < n; base += kMaxTensors) {
    TensorTable tab;
    tab.count = std::min(kMaxTensors, n - base);
    long total = 0;
    for (int i = 0; i < tab.count; ++i) {
      auto& t = tensors[base + i];
      TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                      t.is_contiguous(),
                  "multi_tensor_scale expects contiguous fp32 CUDA tensors");
      tab.t[i] = t.data_ptr<float>();
      tab.n[i] = t.numel();
      total += tab.n[i];
    }
    hipLaunchKernelGGL(mt_scale_kernel, dim3(grid_for(total)), dim3(kBlock), 0,
                       stream, tab, factor_ptr, factor_imm,
                       factor_ptr != nullptr ? 1 : 0);
    HIP_CHECK_LAST();
  }
}

void multi_tensor_scale(std::vector<torch::Tensor> tensors, double factor) {
  mt_scale_impl(tensors, nullptr, (float)factor);
}

void multi_tensor_scale_tensor(std::vector<torch::Tensor> tensors,
                               torch::Tensor factor) {
  TORCH_CHECK(factor.is_cuda() && factor.scalar_type() == torch::kFloat32,
              "factor must be a CUDA fp32 scalar tensor");
  mt_scale_impl(tensors, factor.data_ptr<float>(), 1.f);
}

#endif  // VITFSDP_KERNELS_ONLY
