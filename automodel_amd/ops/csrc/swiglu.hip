// Fused SwiGLU elementwise kernels: y = silu(g) * u, plus backward.
// Memory-bound; bf16x8 vector loads, grid-stride (guide Guideline 11/13).

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

__device__ __forceinline__ float sigmoidf_(float x) { return 1.f / (1.f + __expf(-x)); }

__global__ void swiglu_fwd_kernel(const bf16* __restrict__ g, const bf16* __restrict__ u,
                                  bf16* __restrict__ y, long n8) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 gv = reinterpret_cast<const bf16x8*>(g)[i];
    bf16x8 uv = reinterpret_cast<const bf16x8*>(u)[i];
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(gv.v[j]);
      out.v[j] = f2bf(gf * sigmoidf_(gf) * bf2f(uv.v[j]));
    }
    reinterpret_cast<bf16x8*>(y)[i] = out;
  }
}

__global__ void swiglu_bwd_kernel(const bf16* __restrict__ dy, const bf16* __restrict__ g,
                                  const bf16* __restrict__ u, bf16* __restrict__ dg,
                                  bf16* __restrict__ du, long n8) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 dv = reinterpret_cast<const bf16x8*>(dy)[i];
    bf16x8 gv = reinterpret_cast<const bf16x8*>(g)[i];
    bf16x8 uv = reinterpret_cast<const bf16x8*>(u)[i];
    bf16x8 dgo, duo;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float d = bf2f(dv.v[j]);
      float gf = bf2f(gv.v[j]);
      float uf = bf2f(uv.v[j]);
      float sg = sigmoidf_(gf);
      float silu = gf * sg;
      dgo.v[j] = f2bf(d * uf * sg * (1.f + gf * (1.f - sg)));
      duo.v[j] = f2bf(d * silu);
    }
    reinterpret_cast<bf16x8*>(dg)[i] = dgo;
    reinterpret_cast<bf16x8*>(du)[i] = duo;
  }
}

at::Tensor swiglu_fwd(const at::Tensor& g, const at::Tensor& u) {
  TORCH_CHECK(g.is_cuda() && g.scalar_type() == at::kBFloat16, "swiglu: bf16 CUDA only");
  TORCH_CHECK(g.numel() % 8 == 0, "swiglu: numel must be a multiple of 8");
  auto y = at::empty_like(g);
  const long n8 = g.numel() / 8;
  const int block = 256;
  const int grid = (int)std::min<long>((n8 + block - 1) / block, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(block), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(g.data_ptr()),
                     reinterpret_cast<const bf16*>(u.data_ptr()),
                     reinterpret_cast<bf16*>(y.data_ptr()), n8);
  HIP_CHECK_KERNEL();
  return y;
}

std::tuple<at::Tensor, at::Tensor> swiglu_bwd(const at::Tensor& dy, const at::Tensor& g,
                                              const at::Tensor& u) {
  auto dg = at::empty_like(g);
  auto du = at::empty_like(u);
  const long n8 = g.numel() / 8;
  const int block = 256;
  const int grid = (int)std::min<long>((n8 + block - 1) / block, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(block), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(dy.data_ptr()),
                     reinterpret_cast<const bf16*>(g.data_ptr()),
                     reinterpret_cast<const bf16*>(u.data_ptr()),
                     reinterpret_cast<bf16*>(dg.data_ptr()),
                     reinterpret_cast<bf16*>(du.data_ptr()), n8);
  HIP_CHECK_KERNEL();
  return {dg, du};
}

// Variants over a CONCATENATED [T, 2I] gate|up tensor (fused gate_up_proj
// GEMM output): y[t, i] = silu(gu[t, i]) * gu[t, I + i].
__global__ void swiglu_cat_fwd_kernel(const bf16* __restrict__ gu, bf16* __restrict__ y,
                                      long T, int I) {
  const long t = blockIdx.y;
  const bf16* g = gu + t * (long)(2 * I);
  const bf16* u = g + I;
  bf16* yr = y + t * (long)I;
  for (int i = (blockIdx.x * blockDim.x + threadIdx.x) * 8; i < I;
       i += gridDim.x * blockDim.x * 8) {
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(g + i);
    bf16x8 uv = *reinterpret_cast<const bf16x8*>(u + i);
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(gv.v[j]);
      out.v[j] = f2bf(gf * sigmoidf_(gf) * bf2f(uv.v[j]));
    }
    *reinterpret_cast<bf16x8*>(yr + i) = out;
  }
}

__global__ void swiglu_cat_bwd_kernel(const bf16* __restrict__ dy,
                                      const bf16* __restrict__ gu,
                                      bf16* __restrict__ dgu, long T, int I) {
  const long t = blockIdx.y;
  const bf16* g = gu + t * (long)(2 * I);
  const bf16* u = g + I;
  const bf16* d = dy + t * (long)I;
  bf16* dg = dgu + t * (long)(2 * I);
  bf16* du = dg + I;
  for (int i = (blockIdx.x * blockDim.x + threadIdx.x) * 8; i < I;
       i += gridDim.x * blockDim.x * 8) {
    bf16x8 dv = *reinterpret_cast<const bf16x8*>(d + i);
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(g + i);
    bf16x8 uv = *reinterpret_cast<const bf16x8*>(u + i);
    bf16x8 dgo, duo;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float dd = bf2f(dv.v[j]);
      float gf = bf2f(gv.v[j]);
      float uf = bf2f(uv.v[j]);
      float sg = sigmoidf_(gf);
      dgo.v[j] = f2bf(dd * uf * sg * (1.f + gf * (1.f - sg)));
      duo.v[j] = f2bf(dd * gf * sg);
    }
    *reinterpret_cast<bf16x8*>(dg + i) = dgo;
    *reinterpret_cast<bf16x8*>(du + i) = duo;
  }
}

at::Tensor swiglu_cat_fwd(const at::Tensor& gu) {
  TORCH_CHECK(gu.is_cuda() && gu.scalar_type() == at::kBFloat16 && gu.dim() == 2,
              "swiglu_cat: [T, 2I] bf16");
  const long T = gu.size(0);
  const int I = gu.size(1) / 2;
  TORCH_CHECK(I % 8 == 0, "swiglu_cat: I % 8 == 0");
  auto y = at::empty({T, (long)I}, gu.options());
  const int block = 256;
  const int gx = std::min(32, (I / 8 + block - 1) / block);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_cat_fwd_kernel, dim3(gx, T), dim3(block), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(gu.data_ptr()),
                     reinterpret_cast<bf16*>(y.data_ptr()), T, I);
  HIP_CHECK_KERNEL();
  return y;
}

at::Tensor swiglu_cat_bwd(const at::Tensor& dy, const at::Tensor& gu) {
  const long T = gu.size(0);
  const int I = gu.size(1) / 2;
  auto dgu = at::empty_like(gu);
  const int block = 256;
  const int gx = std::min(32, (I / 8 + block - 1) / block);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_cat_bwd_kernel, dim3(gx, T), dim3(block), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(dy.data_ptr()),
                     reinterpret_cast<const bf16*>(gu.data_ptr()),
                     reinterpret_cast<bf16*>(dgu.data_ptr()), T, I);
  HIP_CHECK_KERNEL();
  return dgu;
}

}  // namespace amd_ops
