// RMSNorm forward + backward for bf16 rows (CDNA4).
// Memory-bound: one workgroup per row, bf16x8 vector loads, x staged in LDS
// so the normalize pass doesn't re-read HBM (guide Appendix B / Guideline 13).
// Replaces TE/QuACK/Liger rms_norm backends of the reference
// (nemo_automodel/components/models/common/utils.py:282).

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

// ---------------------------------------------------------------- forward
// x: [T, H] bf16, w: [H] bf16 -> y: [T, H] bf16, invrms: [T] f32
__global__ void rms_norm_fwd_kernel(const bf16* __restrict__ x,
                                    const bf16* __restrict__ w,
                                    bf16* __restrict__ y,
                                    float* __restrict__ invrms,
                                    int H, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  bf16* xs = reinterpret_cast<bf16*>(smem_raw);                  // H bf16
  float* red = reinterpret_cast<float*>(smem_raw + ((2 * H + 15) & ~15));

  const long row = blockIdx.x;
  const bf16* xr = x + row * (long)H;
  bf16* yr = y + row * (long)H;

  float ss = 0.f;
  for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xr + i);
    *reinterpret_cast<bf16x8*>(xs + i) = xv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(xv.v[j]);
      ss = fmaf(f, f, ss);
    }
  }
  __syncthreads();
  ss = block_reduce_sum(ss, red);
  const float inv = rsqrtf(ss / (float)H + eps);
  if (threadIdx.x == 0) invrms[row] = inv;

  for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xs + i);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      // match torch semantics: round x*inv to bf16, then bf16-multiply by w
      float xn = bf2f(f2bf(bf2f(xv.v[j]) * inv));
      out.v[j] = f2bf(xn * bf2f(wv.v[j]));
    }
    *reinterpret_cast<bf16x8*>(yr + i) = out;
  }
}

// ---------------------------------------------------------------- backward
// dx_i = inv * dy_i*w_i - x_i * inv^3 / H * sum_j(dy_j*w_j*x_j)
// dw_i = sum_rows dy_i * (x_i * inv)
// One block walks rows with stride gridDim.x; each thread owns a fixed set of
// columns so dw accumulates in registers, one atomicAdd per column at the end.
template <int ITERS>
__global__ void rms_norm_bwd_kernel(const bf16* __restrict__ dy,
                                    const bf16* __restrict__ x,
                                    const bf16* __restrict__ w,
                                    const float* __restrict__ invrms,
                                    bf16* __restrict__ dx,
                                    float* __restrict__ dw,
                                    int T, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* red = reinterpret_cast<float*>(smem_raw);

  float dw_acc[ITERS][8];
#pragma unroll
  for (int it = 0; it < ITERS; ++it)
#pragma unroll
    for (int j = 0; j < 8; ++j) dw_acc[it][j] = 0.f;

  // preload w fragments this thread owns
  float wv[ITERS][8];
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int i = (threadIdx.x + it * blockDim.x) * 8;
    if (i < H) {
      bf16x8 t = *reinterpret_cast<const bf16x8*>(w + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) wv[it][j] = bf2f(t.v[j]);
    }
  }

  for (long row = blockIdx.x; row < T; row += gridDim.x) {
    const bf16* xr = x + row * (long)H;
    const bf16* dyr = dy + row * (long)H;
    bf16* dxr = dx + row * (long)H;
    const float inv = invrms[row];

    float xv[ITERS][8], dyv[ITERS][8];
    float s = 0.f;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int i = (threadIdx.x + it * blockDim.x) * 8;
      if (i < H) {
        bf16x8 xt = *reinterpret_cast<const bf16x8*>(xr + i);
        bf16x8 dt = *reinterpret_cast<const bf16x8*>(dyr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          xv[it][j] = bf2f(xt.v[j]);
          dyv[it][j] = bf2f(dt.v[j]);
          s = fmaf(dyv[it][j] * wv[it][j], xv[it][j], s);
        }
      }
    }
    s = block_reduce_sum(s, red);
    const float c = s * inv * inv * inv / (float)H;

#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int i = (threadIdx.x + it * blockDim.x) * 8;
      if (i < H) {
        bf16x8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          out.v[j] = f2bf(inv * dyv[it][j] * wv[it][j] - xv[it][j] * c);
          dw_acc[it][j] = fmaf(dyv[it][j], xv[it][j] * inv, dw_acc[it][j]);
        }
        *reinterpret_cast<bf16x8*>(dxr + i) = out;
      }
    }
  }

#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int i = (threadIdx.x + it * blockDim.x) * 8;
    if (i < H) {
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(dw + i + j, dw_acc[it][j]);
    }
  }
}

std::tuple<at::Tensor, at::Tensor> rms_norm_fwd(const at::Tensor& x,
                                                const at::Tensor& w,
                                                double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.scalar_type() == at::kBFloat16,
              "rms_norm_fwd: x must be 2-D bf16 CUDA");
  const int T = x.size(0), H = x.size(1);
  TORCH_CHECK(H % 8 == 0, "rms_norm_fwd: H must be a multiple of 8, got ", H);
  auto y = at::empty_like(x);
  auto invrms = at::empty({T}, x.options().dtype(at::kFloat));
  const int block = 256;
  const size_t smem = ((2 * H + 15) & ~15) + (block / WAVE_SIZE) * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rms_norm_fwd_kernel, dim3(T), dim3(block), smem, stream.stream(),
                     reinterpret_cast<const bf16*>(x.data_ptr()),
                     reinterpret_cast<const bf16*>(w.data_ptr()),
                     reinterpret_cast<bf16*>(y.data_ptr()),
                     invrms.data_ptr<float>(), H, (float)eps);
  HIP_CHECK_KERNEL();
  return {y, invrms};
}

std::tuple<at::Tensor, at::Tensor> rms_norm_bwd(const at::Tensor& dy,
                                                const at::Tensor& x,
                                                const at::Tensor& w,
                                                const at::Tensor& invrms) {
  const int T = x.size(0), H = x.size(1);
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  const int block = 256;
  const int grid = std::min<long>(T, 1024);
  const size_t smem = (block / WAVE_SIZE) * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream();
  const int iters = ceil_div_i(H, block * 8);

#define LAUNCH_BWD(N)                                                        \
  hipLaunchKernelGGL(rms_norm_bwd_kernel<N>, dim3(grid), dim3(block), smem,  \
                     stream.stream(),                                        \
                     reinterpret_cast<const bf16*>(dy.data_ptr()),           \
                     reinterpret_cast<const bf16*>(x.data_ptr()),            \
                     reinterpret_cast<const bf16*>(w.data_ptr()),            \
                     invrms.data_ptr<float>(),                               \
                     reinterpret_cast<bf16*>(dx.data_ptr()),                 \
                     dw.data_ptr<float>(), T, H)

  if (iters <= 1) LAUNCH_BWD(1);
  else if (iters <= 2) LAUNCH_BWD(2);
  else if (iters <= 4) LAUNCH_BWD(4);
  else if (iters <= 8) LAUNCH_BWD(8);
  else TORCH_CHECK(false, "rms_norm_bwd: H too large: ", H);
#undef LAUNCH_BWD
  HIP_CHECK_KERNEL();
  return {dx, dw};
}

}  // namespace amd_ops
