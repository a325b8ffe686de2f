// Cross-entropy epilogue kernels over per-chunk bf16 logits (CDNA4).
//
// The linear-CE path (loss/linear_ce.py "hybrid" backend) computes per-chunk
// logits with a hipBLASLt GEMM and hands them to these kernels:
//   ce_fwd_logits: one pass over each row -> online (max, sumexp), writes
//     lse[t] and atomically accumulates sum loss. No f32 logits copy, no
//     softmax materialization (replaces a ~550 ms/step elementwise chain of
//     torch ops measured in profiles/bench8b round-1).
//   ce_bwd_logits: overwrites logits IN PLACE with d(logits) =
//     (softmax - onehot(label)) * dloss — single bf16 read+write pass.
//
// Together with the GEMMs this is the MI355X equivalent of cut-cross-entropy
// (reference loss/linear_ce.py:253): the full [T, V] logits tensor never
// exists, only one [chunk, V] bf16 buffer.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

// combine two (m, s) online-softmax states
__device__ __forceinline__ void combine_ms(float& m, float& s, float m2, float s2) {
  float M = fmaxf(m, m2);
  s = s * __expf(m - M) + s2 * __expf(m2 - M);
  m = M;
}

// one block per row; threads stride over V with bf16x8 loads.
__global__ void ce_fwd_logits_kernel(const bf16* __restrict__ logits,
                                     const long* __restrict__ labels,
                                     float* __restrict__ lse,
                                     float* __restrict__ loss_sum,
                                     int V, long ignore_index) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);  // 2 floats per wave

  const long row = blockIdx.x;
  const bf16* x = logits + row * (long)V;
  float m = -1e30f, s = 0.f;
  const int stride = blockDim.x * 8;
  for (int i = threadIdx.x * 8; i + 7 < V; i += stride) {  // V % 8 == 0 enforced
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(xv.v[j]);
      if (f > m) { s *= __expf(m - f); m = f; }
      s += __expf(f - m);
    }
  }

  // wave reduce
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    combine_ms(m, s, __shfl_xor(m, off), __shfl_xor(s, off));
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  if ((threadIdx.x & 63) == 0) { red[2 * wid] = m; red[2 * wid + 1] = s; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = red[0], S = red[1];
    for (int w = 1; w < nwaves; ++w) combine_ms(M, S, red[2 * w], red[2 * w + 1]);
    const float l = M + __logf(S);
    lse[row] = l;
    const long y = labels[row];
    if (y != ignore_index) {
      atomicAdd(loss_sum, l - bf2f(x[y]));
    }
  }
}

// per-row online (max, sumexp) stats only — vocab-parallel CE exchanges
// (m, d, X_y) across TP ranks (reference loss/te_parallel_ce.py:45-191).
__global__ void ce_stats_logits_kernel(const bf16* __restrict__ logits,
                                       float* __restrict__ m_out,
                                       float* __restrict__ s_out, int V) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);
  const long row = blockIdx.x;
  const bf16* x = logits + row * (long)V;
  float m = -1e30f, s = 0.f;
  const int stride = blockDim.x * 8;
  for (int i = threadIdx.x * 8; i + 7 < V; i += stride) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(xv.v[j]);
      if (f > m) { s *= __expf(m - f); m = f; }
      s += __expf(f - m);
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    combine_ms(m, s, __shfl_xor(m, off), __shfl_xor(s, off));
  const int wid = threadIdx.x / WAVE_SIZE;
  if ((threadIdx.x & 63) == 0) { red[2 * wid] = m; red[2 * wid + 1] = s; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = red[0], S = red[1];
    for (int w = 1; w < blockDim.x / WAVE_SIZE; ++w)
      combine_ms(M, S, red[2 * w], red[2 * w + 1]);
    m_out[row] = M;
    s_out[row] = S;
  }
}

// grad in place: g = (exp(x - lse) - onehot) * dloss ; ignored rows -> 0.
__global__ void ce_bwd_logits_kernel(bf16* __restrict__ logits,
                                     const long* __restrict__ labels,
                                     const float* __restrict__ lse,
                                     const float* __restrict__ dloss,
                                     int V, long ignore_index, long vocab_offset) {
  const long row = blockIdx.y;
  const long y_raw = labels[row];
  const long y = (y_raw == ignore_index) ? ignore_index : y_raw - vocab_offset;
  const float d = dloss[0];
  const float l = lse[row];
  bf16* x = logits + row * (long)V;
  const bool ignored = (y_raw == ignore_index);
  for (int i = (blockIdx.x * blockDim.x + threadIdx.x) * 8; i < V;
       i += gridDim.x * blockDim.x * 8) {
    bf16x8 xv = *reinterpret_cast<bf16x8*>(x + i);
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (ignored) { out.v[j] = f2bf(0.f); continue; }
      float p = __expf(bf2f(xv.v[j]) - l);
      if ((long)(i + j) == y) p -= 1.f;
      out.v[j] = f2bf(p * d);
    }
    *reinterpret_cast<bf16x8*>(x + i) = out;
  }
}

std::tuple<at::Tensor, at::Tensor> ce_fwd_logits(const at::Tensor& logits,
                                                 const at::Tensor& labels,
                                                 at::Tensor loss_sum) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.scalar_type() == at::kBFloat16,
              "ce_fwd_logits: logits must be [T,V] bf16");
  TORCH_CHECK(labels.scalar_type() == at::kLong, "labels must be int64");
  const long T = logits.size(0);
  const int V = logits.size(1);
  TORCH_CHECK(V % 8 == 0, "ce_fwd_logits: V must be a multiple of 8, got ", V);
  auto lse = at::empty({T}, logits.options().dtype(at::kFloat));
  const int block = 512;
  const size_t smem = (block / WAVE_SIZE) * 2 * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(ce_fwd_logits_kernel, dim3(T), dim3(block), smem, stream.stream(),
                     reinterpret_cast<const bf16*>(logits.data_ptr()),
                     labels.data_ptr<long>(), lse.data_ptr<float>(),
                     loss_sum.data_ptr<float>(), V, -100);
  HIP_CHECK_KERNEL();
  return {lse, loss_sum};
}

void ce_bwd_logits(at::Tensor logits, const at::Tensor& labels, const at::Tensor& lse,
                   const at::Tensor& dloss, int64_t vocab_offset) {
  const long T = logits.size(0);
  const int V = logits.size(1);
  TORCH_CHECK(V % 8 == 0, "ce_bwd_logits: V must be a multiple of 8");
  const int block = 256;
  const int gx = std::min(64, (V / 8 + block - 1) / block);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(ce_bwd_logits_kernel, dim3(gx, T), dim3(block), 0, stream.stream(),
                     reinterpret_cast<bf16*>(logits.data_ptr()),
                     labels.data_ptr<long>(), lse.data_ptr<float>(),
                     dloss.data_ptr<float>(), V, -100, vocab_offset);
  HIP_CHECK_KERNEL();
}

std::tuple<at::Tensor, at::Tensor> ce_stats_logits(const at::Tensor& logits) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 &&
              logits.scalar_type() == at::kBFloat16, "ce_stats_logits: [T,V] bf16");
  const long T = logits.size(0);
  const int V = logits.size(1);
  TORCH_CHECK(V % 8 == 0, "ce_stats_logits: V must be a multiple of 8");
  auto m = at::empty({T}, logits.options().dtype(at::kFloat));
  auto s = at::empty({T}, logits.options().dtype(at::kFloat));
  const int block = 512;
  const size_t smem = (block / WAVE_SIZE) * 2 * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(ce_stats_logits_kernel, dim3(T), dim3(block), smem, stream.stream(),
                     reinterpret_cast<const bf16*>(logits.data_ptr()),
                     m.data_ptr<float>(), s.data_ptr<float>(), V);
  HIP_CHECK_KERNEL();
  return {m, s};
}

}  // namespace amd_ops
