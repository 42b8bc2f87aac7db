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
//
// Gradients may arrive in bf16 (the engine's comm-dtype reduced shards,
// fsdp.py _finalize_unit): the kernels upcast at the register boundary
// and fold the pending 1/world_size prescale plus the deferred clip
// coefficient into the same read — the bf16 grad shard is the ONLY
// gradient memory traffic of the whole step.

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
  const void* g[kMaxTensors];  // fp32 or bf16 (GRAD_BF16 template flag)
  float* m[kMaxTensors];
  float* v[kMaxTensors];
  unsigned short* mir[kMaxTensors];  // optional bf16 mirror of p
  long n[kMaxTensors];
  int count;
};

struct TensorTable {
  void* t[kMaxTensors];  // fp32 or bf16 (per-launch template flag)
  long n[kMaxTensors];
  int count;
};

template <bool GRAD_BF16>
__global__ void fused_adamw_kernel(AdamWTable tab, float lr, float beta1,
                                   float beta2, float eps, float wd,
                                   float bias_c1, float bias_c2,
                                   const float* __restrict__ grad_scale,
                                   float prescale) {
  // deferred gradient clipping + mean divide: the clip coefficient (a
  // device scalar from clip_grad_norm_(defer_scale=True)) and the
  // 1/world_size prescale scale g on the fly, replacing separate full
  // read+write passes over every gradient shard
  const float gs = (grad_scale ? *grad_scale : 1.f) * prescale;
  const float decay = 1.f - lr * wd;
  const float step_size = lr / bias_c1;
  const float inv_sqrt_c2 = rsqrtf(bias_c2);
  for (int ti = 0; ti < tab.count; ++ti) {
    float4* p = (float4*)tab.p[ti];
    float4* m = (float4*)tab.m[ti];
    float4* v = (float4*)tab.v[ti];
    const float4* gf = (const float4*)tab.g[ti];
    const ushort4* gb = (const ushort4*)tab.g[ti];
    const long n4 = tab.n[ti] / 4;
    for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n4;
         i += (long)gridDim.x * kBlock) {
      float4 pv = p[i], mv = m[i], vv = v[i];
      float gj4[4];
      if (GRAD_BF16) {
        ushort4 gv = gb[i];
#pragma unroll
        for (int j = 0; j < 4; ++j) gj4[j] = bf16_to_f32((&gv.x)[j]);
      } else {
        float4 gv = gf[i];
#pragma unroll
        for (int j = 0; j < 4; ++j) gj4[j] = (&gv.x)[j];
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float pj = (&pv.x)[j], gj = gj4[j] * gs;
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
      float gj = GRAD_BF16 ? bf16_to_f32(((const unsigned short*)tab.g[ti])[i])
                           : ((const float*)tab.g[ti])[i];
      gj *= gs;
      float pj = tab.p[ti][i];
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

template <bool BF16>
__global__ void mt_sqnorm_kernel(TensorTable tab, float* __restrict__ out) {
  __shared__ float scratch[kBlock / WAVE_SIZE];
  float acc = 0.f;
  for (int ti = 0; ti < tab.count; ++ti) {
    if (BF16) {
      // 8 bf16 per 16 B vector load
      const ushort8_t* t = (const ushort8_t*)tab.t[ti];
      const long n8 = tab.n[ti] / 8;
      for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n8;
           i += (long)gridDim.x * kBlock) {
        ushort8_t u = t[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float v = bf16_to_f32(u[j]);
          acc += v * v;
        }
      }
      const long tail = tab.n[ti] & 7;
      if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
        float v = bf16_to_f32(
            ((const unsigned short*)tab.t[ti])[tab.n[ti] - tail + threadIdx.x]);
        acc += v * v;
      }
    } else {
      const float4* t = (const float4*)tab.t[ti];
      const long n4 = tab.n[ti] / 4;
      for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n4;
           i += (long)gridDim.x * kBlock) {
        float4 v = t[i];
        acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
      }
      const long tail = tab.n[ti] & 3;
      if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
        float v = ((const float*)tab.t[ti])[tab.n[ti] - tail + threadIdx.x];
        acc += v * v;
      }
    }
  }
  acc = block_sum<kBlock / WAVE_SIZE>(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

template <bool BF16>
__global__ void mt_scale_kernel(TensorTable tab,
                                const float* __restrict__ factor_ptr,
                                float factor_imm, int use_ptr) {
  const float f = use_ptr ? *factor_ptr : factor_imm;
  for (int ti = 0; ti < tab.count; ++ti) {
    if (BF16) {
      ushort8_t* t = (ushort8_t*)tab.t[ti];
      const long n8 = tab.n[ti] / 8;
      for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n8;
           i += (long)gridDim.x * kBlock) {
        ushort8_t u = t[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) u[j] = f32_to_bf16(bf16_to_f32(u[j]) * f);
        t[i] = u;
      }
      const long tail = tab.n[ti] & 7;
      if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
        unsigned short* s =
            (unsigned short*)tab.t[ti] + (tab.n[ti] - tail + threadIdx.x);
        *s = f32_to_bf16(bf16_to_f32(*s) * f);
      }
    } else {
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
        ((float*)tab.t[ti])[tab.n[ti] - tail + threadIdx.x] *= f;
      }
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
                 c10::optional<torch::Tensor> grad_scale,
                 double grad_prescale) {
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
  if (n == 0) return;
  const auto grad_dtype = grads[0].scalar_type();
  TORCH_CHECK(grad_dtype == torch::kFloat32 || grad_dtype == torch::kBFloat16,
              "fused_adamw: grads must be fp32 or bf16");
  for (int base = 0; base < n; base += kMaxTensors) {
    AdamWTable tab;
    tab.count = std::min(kMaxTensors, n - base);
    long total = 0;
    for (int i = 0; i < tab.count; ++i) {
      auto& p = params[base + i];
      auto& g = grads[base + i];
      TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kFloat32 &&
                      p.is_contiguous(),
                  "fused_adamw expects contiguous fp32 CUDA tensors");
      TORCH_CHECK(g.scalar_type() == grad_dtype && g.is_contiguous() &&
                      g.numel() >= p.numel(),
                  "fused_adamw: grads must share one dtype and cover params");
      tab.p[i] = p.data_ptr<float>();
      tab.g[i] = g.data_ptr();
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
    if (grad_dtype == torch::kBFloat16) {
      hipLaunchKernelGGL(fused_adamw_kernel<true>, dim3(grid_for(total)),
                         dim3(kBlock), 0, stream, tab, (float)lr, (float)beta1,
                         (float)beta2, (float)eps, (float)weight_decay,
                         (float)bias_c1, (float)bias_c2, gs_ptr,
                         (float)grad_prescale);
    } else {
      hipLaunchKernelGGL(fused_adamw_kernel<false>, dim3(grid_for(total)),
                         dim3(kBlock), 0, stream, tab, (float)lr, (float)beta1,
                         (float)beta2, (float)eps, (float)weight_decay,
                         (float)bias_c1, (float)bias_c2, gs_ptr,
                         (float)grad_prescale);
    }
    HIP_CHECK_LAST();
  }
}

namespace {

// split-by-dtype driver shared by sqnorm and scale: launches the fp32
// and bf16 template instantiations over homogeneous chunk tables
template <typename LaunchF32, typename LaunchBF16>
void mt_dispatch(std::vector<torch::Tensor>& tensors, const char* what,
                 LaunchF32 launch_f32, LaunchBF16 launch_bf16) {
  std::vector<torch::Tensor*> f32s, bf16s;
  for (auto& t : tensors) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous(), what,
                " expects contiguous CUDA tensors");
    if (t.scalar_type() == torch::kFloat32)
      f32s.push_back(&t);
    else if (t.scalar_type() == torch::kBFloat16)
      bf16s.push_back(&t);
    else
      TORCH_CHECK(false, what, ": fp32 or bf16 only");
  }
  for (int pass = 0; pass < 2; ++pass) {
    auto& list = pass == 0 ? f32s : bf16s;
    const int n = (int)list.size();
    for (int base = 0; base < n; base += kMaxTensors) {
      TensorTable tab;
      tab.count = std::min(kMaxTensors, n - base);
      long total = 0;
      for (int i = 0; i < tab.count; ++i) {
        tab.t[i] = list[base + i]->data_ptr();
        tab.n[i] = list[base + i]->numel();
        total += tab.n[i];
      }
      if (pass == 0)
        launch_f32(tab, grid_for(total));
      else
        launch_bf16(tab, grid_for(total));
      HIP_CHECK_LAST();
    }
  }
}

}  // namespace

torch::Tensor multi_tensor_sqnorm(std::vector<torch::Tensor> tensors) {
  auto stream = at::cuda::getCurrentCUDAStream();
  auto out = torch::zeros({}, tensors[0].options().dtype(torch::kFloat32));
  float* out_ptr = out.data_ptr<float>();
  mt_dispatch(
      tensors, "multi_tensor_sqnorm",
      [&](const TensorTable& tab, int grid) {
        hipLaunchKernelGGL(mt_sqnorm_kernel<false>, dim3(grid), dim3(kBlock),
                           0, stream, tab, out_ptr);
      },
      [&](const TensorTable& tab, int grid) {
        hipLaunchKernelGGL(mt_sqnorm_kernel<true>, dim3(grid), dim3(kBlock),
                           0, stream, tab, out_ptr);
      });
  return out;
}

static void mt_scale_impl(std::vector<torch::Tensor>& tensors,
                          const float* factor_ptr, float factor_imm) {
  auto stream = at::cuda::getCurrentCUDAStream();
  const int use_ptr = factor_ptr != nullptr ? 1 : 0;
  mt_dispatch(
      tensors, "multi_tensor_scale",
      [&](const TensorTable& tab, int grid) {
        hipLaunchKernelGGL(mt_scale_kernel<false>, dim3(grid), dim3(kBlock),
                           0, stream, tab, factor_ptr, factor_imm, use_ptr);
      },
      [&](const TensorTable& tab, int grid) {
        hipLaunchKernelGGL(mt_scale_kernel<true>, dim3(grid), dim3(kBlock),
                           0, stream, tab, factor_ptr, factor_imm, use_ptr);
      });
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
