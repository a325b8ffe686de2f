// Skinny-batch bf16 GEMV for KV-cached decode: y[B,N] = x[B,K] @ W[N,K]^T.
//
// Decode-step linears are m<=16 GEMMs — pure weight-streaming. hipBLASLt's
// GEMV path measures ~23% of HBM bandwidth on these shapes (profiled in
// profiles/decode_*.json); this kernel streams each W row once with b128
// loads (guide Guideline 13) and broadcasts the tiny x through L2:
//   * one wave per output row, lanes cover K in 16-byte chunks
//     (lane l reads W[row, l*8 + i*512 .. +8) — 1 KB/instruction/wave,
//     fully coalesced);
//   * batch columns (B) accumulate in registers in the same pass, so W
//     traffic is independent of B;
//   * fp32 dot accumulation + wave reduce, lane 0 stores.
// Grid = N/4 workgroups of 4 waves: N=4096 -> 1024 blocks >> 256 CUs.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

// Two rows per wave: doubles outstanding b128 loads per wave (latency
// hiding for the short 8-iteration K sweep at K=4096).
template <int B>
__global__ __launch_bounds__(256, 4) void gemv_bf16_kernel(
    const bf16* __restrict__ W, const bf16* __restrict__ x,
    const bf16* __restrict__ bias, bf16* __restrict__ y, int N, int K) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = blockIdx.x * (blockDim.x / WAVE_SIZE) + (threadIdx.x / WAVE_SIZE);
  const int row0 = 2 * wid;
  if (row0 >= N) return;
  const bool two = row0 + 1 < N;
  float acc0[B], acc1[B];
#pragma unroll
  for (int b = 0; b < B; ++b) acc0[b] = acc1[b] = 0.f;
  const bf16* wrow0 = W + (long)row0 * K;
  const bf16* wrow1 = wrow0 + (two ? K : 0);
  for (int k = lane * 8; k < K; k += WAVE_SIZE * 8) {
    const bf16x8 w0 = *reinterpret_cast<const bf16x8*>(wrow0 + k);
    const bf16x8 w1 = *reinterpret_cast<const bf16x8*>(wrow1 + k);
#pragma unroll
    for (int b = 0; b < B; ++b) {
      const bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + (long)b * K + k);
      float d0 = 0.f, d1 = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xf = bf2f(xv.v[j]);
        d0 += bf2f(w0.v[j]) * xf;
        d1 += bf2f(w1.v[j]) * xf;
      }
      acc0[b] += d0;
      acc1[b] += d1;
    }
  }
#pragma unroll
  for (int b = 0; b < B; ++b) {
    const float t0 = wave_reduce_sum(acc0[b]);
    const float t1 = wave_reduce_sum(acc1[b]);
    if (lane == 0) {
      y[(long)b * N + row0] = f2bf(t0 + (bias ? bf2f(bias[row0]) : 0.f));
      if (two)
        y[(long)b * N + row0 + 1] = f2bf(t1 + (bias ? bf2f(bias[row0 + 1]) : 0.f));
    }
  }
}

at::Tensor gemv_bf16(const at::Tensor& x, const at::Tensor& w,
                     const c10::optional<at::Tensor>& bias) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda(), "gemv_bf16: GPU tensors required");
  TORCH_CHECK(x.dtype() == at::kBFloat16 && w.dtype() == at::kBFloat16,
              "gemv_bf16: bf16 only");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "gemv_bf16: x [B,K], w [N,K]");
  const int B = (int)x.size(0), K = (int)x.size(1), N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  TORCH_CHECK(B >= 1 && B <= 16, "gemv_bf16: B in [1,16], got ", B,
              " (larger batches belong on hipBLASLt)");
  TORCH_CHECK(K % (WAVE_SIZE * 8) == 0, "gemv_bf16: K % 512 == 0, got ", K);
  auto xc = x.contiguous(), wc = w.contiguous();
  auto y = at::empty({B, N}, x.options());
  const bf16* bptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->numel() == N && bias->dtype() == at::kBFloat16, "bias [N] bf16");
    bptr = reinterpret_cast<const bf16*>(bias->contiguous().data_ptr());
  }
  const int waves_per_block = 4;
  const int rows_per_block = 2 * waves_per_block;
  const dim3 grid((N + rows_per_block - 1) / rows_per_block);
  auto stream = c10::hip::getCurrentHIPStream();
#define LAUNCH(BB)                                                             \
  gemv_bf16_kernel<BB><<<grid, dim3(waves_per_block * WAVE_SIZE), 0, stream>>>( \
      reinterpret_cast<const bf16*>(wc.data_ptr()),                            \
      reinterpret_cast<const bf16*>(xc.data_ptr()), bptr,                      \
      reinterpret_cast<bf16*>(y.data_ptr()), N, K)
  switch (B) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 3: LAUNCH(3); break;
    case 4: LAUNCH(4); break;
    case 8: LAUNCH(8); break;
    case 16: LAUNCH(16); break;
    default: TORCH_CHECK(false, "gemv_bf16: pad B to one of 1,2,3,4,8,16 (python wrapper does)");
  }
#undef LAUNCH
  HIP_CHECK_KERNEL();
  return y;
}

}  // namespace amd_ops
