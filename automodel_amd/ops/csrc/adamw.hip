// Fused AdamW step (CDNA4): bf16 params + fp32 master weights + fp32 moments.
// Semantics match torch.optim.AdamW exactly (decoupled weight decay applied
// multiplicatively before the moment update) so the optimizer parity test can
// compare against torch. Replaces the reference's Apex FusedAdam path
// (nemo_automodel/components/optim/optimizer.py:297).
//
// Memory-bound: ~26 B read + 22 B write per element; float4 / bf16x4 vectors.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

__global__ void adamw_bf16_kernel(bf16* __restrict__ p, const bf16* __restrict__ g,
                                  float* __restrict__ master, float* __restrict__ m,
                                  float* __restrict__ v, long n4,
                                  float lr, float b1, float b2, float eps,
                                  float wd_factor, float inv_bc1, float inv_sqrt_bc2) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    bf16x4 gv = reinterpret_cast<const bf16x4*>(g)[i];
    float4 mw = reinterpret_cast<const float4*>(master)[i];
    float4 mv = reinterpret_cast<const float4*>(m)[i];
    float4 vv = reinterpret_cast<const float4*>(v)[i];
    float* mwp = &mw.x; float* mvp = &mv.x; float* vvp = &vv.x;
    bf16x4 pout;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gf = bf2f(gv.v[j]);
      float w = mwp[j] * wd_factor;
      float m_ = b1 * mvp[j] + (1.f - b1) * gf;
      float v_ = b2 * vvp[j] + (1.f - b2) * gf * gf;
      float denom = sqrtf(v_) * inv_sqrt_bc2 + eps;
      w -= lr * inv_bc1 * m_ / denom;
      mwp[j] = w; mvp[j] = m_; vvp[j] = v_;
      pout.v[j] = f2bf(w);
    }
    reinterpret_cast<float4*>(master)[i] = mw;
    reinterpret_cast<float4*>(m)[i] = mv;
    reinterpret_cast<float4*>(v)[i] = vv;
    reinterpret_cast<bf16x4*>(p)[i] = pout;
  }
}

// bf16-STATE variant: param/grad/m/v all bf16, NO fp32 master — halves
// optimizer memory for single-GPU 30B-class benches (update math in fp32
// registers; only the stored moments are quantized).
__global__ void adamw_bf16_state_kernel(bf16* __restrict__ p, const bf16* __restrict__ g,
                                        bf16* __restrict__ m, bf16* __restrict__ v,
                                        long n4, float lr, float b1, float b2,
                                        float eps, float wd_factor, float inv_bc1,
                                        float inv_sqrt_bc2) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (long)gridDim.x * blockDim.x) {
    bf16x4 gv = reinterpret_cast<const bf16x4*>(g)[i];
    bf16x4 pv = reinterpret_cast<const bf16x4*>(p)[i];
    bf16x4 mv = reinterpret_cast<const bf16x4*>(m)[i];
    bf16x4 vv = reinterpret_cast<const bf16x4*>(v)[i];
    bf16x4 pout, mout, vout;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gf = bf2f(gv.v[j]);
      float w = bf2f(pv.v[j]) * wd_factor;
      float m_ = b1 * bf2f(mv.v[j]) + (1.f - b1) * gf;
      float v_ = b2 * bf2f(vv.v[j]) + (1.f - b2) * gf * gf;
      float denom = sqrtf(v_) * inv_sqrt_bc2 + eps;
      w -= lr * inv_bc1 * m_ / denom;
      mout.v[j] = f2bf(m_); vout.v[j] = f2bf(v_); pout.v[j] = f2bf(w);
    }
    reinterpret_cast<bf16x4*>(m)[i] = mout;
    reinterpret_cast<bf16x4*>(v)[i] = vout;
    reinterpret_cast<bf16x4*>(p)[i] = pout;
  }
}

// fp32-param variant (no separate master copy); grad may be fp32 too.
__global__ void adamw_f32_kernel(float* __restrict__ p, const float* __restrict__ g,
                                 float* __restrict__ m, float* __restrict__ v, long n,
                                 float lr, float b1, float b2, float eps,
                                 float wd_factor, float inv_bc1, float inv_sqrt_bc2) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float gf = g[i];
    float w = p[i] * wd_factor;
    float m_ = b1 * m[i] + (1.f - b1) * gf;
    float v_ = b2 * v[i] + (1.f - b2) * gf * gf;
    w -= lr * inv_bc1 * m_ / (sqrtf(v_) * inv_sqrt_bc2 + eps);
    p[i] = w; m[i] = m_; v[i] = v_;
  }
}

void adamw_step(at::Tensor param, at::Tensor grad, at::Tensor master, at::Tensor m,
                at::Tensor v, int64_t step, double lr, double beta1, double beta2,
                double eps, double weight_decay) {
  TORCH_CHECK(param.is_cuda(), "adamw_step: CUDA only");
  const double bc1 = 1.0 - std::pow(beta1, (double)step);
  const double bc2 = 1.0 - std::pow(beta2, (double)step);
  const float wd_factor = (float)(1.0 - lr * weight_decay);
  const float inv_bc1 = (float)(1.0 / bc1);
  const float inv_sqrt_bc2 = (float)(1.0 / std::sqrt(bc2));
  auto stream = c10::hip::getCurrentHIPStream();
  const int block = 256;
  if (param.scalar_type() == at::kBFloat16 && m.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(param.numel() % 4 == 0, "adamw_step: numel must be multiple of 4");
    const long n4 = param.numel() / 4;
    const int grid = (int)std::min<long>((n4 + block - 1) / block, 4096);
    hipLaunchKernelGGL(adamw_bf16_state_kernel, dim3(grid), dim3(block), 0,
                       stream.stream(),
                       reinterpret_cast<bf16*>(param.data_ptr()),
                       reinterpret_cast<const bf16*>(grad.data_ptr()),
                       reinterpret_cast<bf16*>(m.data_ptr()),
                       reinterpret_cast<bf16*>(v.data_ptr()),
                       n4, (float)lr, (float)beta1, (float)beta2, (float)eps,
                       wd_factor, inv_bc1, inv_sqrt_bc2);
  } else if (param.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(param.numel() % 4 == 0, "adamw_step: numel must be multiple of 4");
    const long n4 = param.numel() / 4;
    const int grid = (int)std::min<long>((n4 + block - 1) / block, 4096);
    hipLaunchKernelGGL(adamw_bf16_kernel, dim3(grid), dim3(block), 0, stream.stream(),
                       reinterpret_cast<bf16*>(param.data_ptr()),
                       reinterpret_cast<const bf16*>(grad.data_ptr()),
                       master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                       n4, (float)lr, (float)beta1, (float)beta2, (float)eps,
                       wd_factor, inv_bc1, inv_sqrt_bc2);
  } else {
    const long n = param.numel();
    const int grid = (int)std::min<long>((n + block - 1) / block, 4096);
    hipLaunchKernelGGL(adamw_f32_kernel, dim3(grid), dim3(block), 0, stream.stream(),
                       param.data_ptr<float>(), grad.data_ptr<float>(),
                       m.data_ptr<float>(), v.data_ptr<float>(), n,
                       (float)lr, (float)beta1, (float)beta2, (float)eps,
                       wd_factor, inv_bc1, inv_sqrt_bc2);
  }
  HIP_CHECK_KERNEL();
}

}  // namespace amd_ops
