// Soft (distribution-target) cross entropy over bf16 logits (CDNA4).
// KD path — replaces the reference's Triton soft CE
// (nemo_automodel/components/loss/triton/soft_cross_entropy.py).
//
//   fwd (one pass, online): loss_row = lse_s - sum_v softmax(t)[v] * s[v]
//     tracked as (m_s, d_s), (m_t, d_t) and u = sum exp(t - m_t) * s with
//     rescaling on running-max updates.
//   bwd: d(s) = (softmax(s) - softmax(t)) * dloss / T_rows... (caller scales)

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

__global__ void soft_ce_fwd_kernel(const bf16* __restrict__ s_logits,
                                   const bf16* __restrict__ t_logits,
                                   float* __restrict__ lse_s_out,
                                   float* __restrict__ lse_t_out,
                                   float* __restrict__ loss_out, int V) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);  // 5 floats per wave

  const long row = blockIdx.x;
  const bf16* s = s_logits + row * (long)V;
  const bf16* t = t_logits + row * (long)V;
  float ms = -1e30f, ds = 0.f;
  float mt = -1e30f, dt = 0.f, u = 0.f;
  const int stride = blockDim.x * 8;
  for (int i = threadIdx.x * 8; i + 7 < V; i += stride) {
    bf16x8 sv = *reinterpret_cast<const bf16x8*>(s + i);
    bf16x8 tv = *reinterpret_cast<const bf16x8*>(t + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float sf = bf2f(sv.v[j]);
      float tf = bf2f(tv.v[j]);
      if (sf > ms) { ds *= __expf(ms - sf); ms = sf; }
      ds += __expf(sf - ms);
      if (tf > mt) { float r = __expf(mt - tf); dt *= r; u *= r; mt = tf; }
      float e = __expf(tf - mt);
      dt += e;
      u += e * sf;
    }
  }
  // wave reduce (combine 5-tuples)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ms2 = __shfl_xor(ms, off), ds2 = __shfl_xor(ds, off);
    float M = fmaxf(ms, ms2);
    ds = ds * __expf(ms - M) + ds2 * __expf(ms2 - M);
    ms = M;
    float mt2 = __shfl_xor(mt, off), dt2 = __shfl_xor(dt, off), u2 = __shfl_xor(u, off);
    float Mt = fmaxf(mt, mt2);
    float ra = __expf(mt - Mt), rb = __expf(mt2 - Mt);
    dt = dt * ra + dt2 * rb;
    u = u * ra + u2 * rb;
    mt = Mt;
  }
  const int wid = threadIdx.x / WAVE_SIZE;
  if ((threadIdx.x & 63) == 0) {
    red[5 * wid] = ms; red[5 * wid + 1] = ds;
    red[5 * wid + 2] = mt; red[5 * wid + 3] = dt; red[5 * wid + 4] = u;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float Ms = red[0], Ds = red[1], Mt = red[2], Dt = red[3], U = red[4];
    for (int w = 1; w < blockDim.x / WAVE_SIZE; ++w) {
      float ms2 = red[5 * w], ds2 = red[5 * w + 1];
      float M = fmaxf(Ms, ms2);
      Ds = Ds * __expf(Ms - M) + ds2 * __expf(ms2 - M);
      Ms = M;
      float mt2 = red[5 * w + 2], dt2 = red[5 * w + 3], u2 = red[5 * w + 4];
      float M2 = fmaxf(Mt, mt2);
      float ra = __expf(Mt - M2), rb = __expf(mt2 - M2);
      Dt = Dt * ra + dt2 * rb;
      U = U * ra + u2 * rb;
      Mt = M2;
    }
    const float lse_s = Ms + __logf(Ds);
    const float lse_t = Mt + __logf(Dt);
    lse_s_out[row] = lse_s;
    lse_t_out[row] = lse_t;
    loss_out[row] = lse_s - U / Dt;
  }
}

__global__ void soft_ce_bwd_kernel(const bf16* __restrict__ s_logits,
                                   const bf16* __restrict__ t_logits,
                                   const float* __restrict__ lse_s,
                                   const float* __restrict__ lse_t,
                                   const float* __restrict__ dloss,
                                   bf16* __restrict__ grad, int V) {
  const long row = blockIdx.y;
  const float ls = lse_s[row], lt = lse_t[row], d = dloss[0];
  const bf16* s = s_logits + row * (long)V;
  const bf16* t = t_logits + row * (long)V;
  bf16* g = grad + row * (long)V;
  for (int i = (blockIdx.x * blockDim.x + threadIdx.x) * 8; i < V;
       i += gridDim.x * blockDim.x * 8) {
    bf16x8 sv = *reinterpret_cast<const bf16x8*>(s + i);
    bf16x8 tv = *reinterpret_cast<const bf16x8*>(t + i);
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float ps = __expf(bf2f(sv.v[j]) - ls);
      float pt = __expf(bf2f(tv.v[j]) - lt);
      out.v[j] = f2bf((ps - pt) * d);
    }
    *reinterpret_cast<bf16x8*>(g + i) = out;
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> soft_ce_fwd(const at::Tensor& s,
                                                           const at::Tensor& t) {
  TORCH_CHECK(s.is_cuda() && s.dim() == 2 && s.scalar_type() == at::kBFloat16 &&
              s.sizes() == t.sizes(), "soft_ce_fwd: s/t [T,V] bf16");
  const long T = s.size(0);
  const int V = s.size(1);
  TORCH_CHECK(V % 8 == 0, "soft_ce_fwd: V % 8 == 0");
  auto lse_s = at::empty({T}, s.options().dtype(at::kFloat));
  auto lse_t = at::empty({T}, s.options().dtype(at::kFloat));
  auto loss = at::empty({T}, s.options().dtype(at::kFloat));
  const int block = 512;
  const size_t smem = (block / WAVE_SIZE) * 5 * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(soft_ce_fwd_kernel, dim3(T), dim3(block), smem, stream.stream(),
                     reinterpret_cast<const bf16*>(s.data_ptr()),
                     reinterpret_cast<const bf16*>(t.data_ptr()),
                     lse_s.data_ptr<float>(), lse_t.data_ptr<float>(),
                     loss.data_ptr<float>(), V);
  HIP_CHECK_KERNEL();
  return {loss, lse_s, lse_t};
}

at::Tensor soft_ce_bwd(const at::Tensor& s, const at::Tensor& t,
                       const at::Tensor& lse_s, const at::Tensor& lse_t,
                       const at::Tensor& dloss) {
  const long T = s.size(0);
  const int V = s.size(1);
  auto grad = at::empty_like(s);
  const int block = 256;
  const int gx = std::min(64, (V / 8 + block - 1) / block);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(soft_ce_bwd_kernel, dim3(gx, T), dim3(block), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(s.data_ptr()),
                     reinterpret_cast<const bf16*>(t.data_ptr()),
                     lse_s.data_ptr<float>(), lse_t.data_ptr<float>(),
                     dloss.data_ptr<float>(),
                     reinterpret_cast<bf16*>(grad.data_ptr()), V);
  HIP_CHECK_KERNEL();
  return grad;
}

}  // namespace amd_ops
