// NF4 (4-bit NormalFloat) dequantization for QLoRA on MI355X.
//
// Reference behavior: nemo_automodel supports QLoRA via bitsandbytes NF4
// (nemo_automodel/components/_peft/lora.py quantized base weights). Here the
// format is implemented natively: blockwise absmax-scaled 4-bit codes packed
// two per byte, dequantized on the fly to bf16 right before the hipBLASLt
// GEMM. The kernel is pure-bandwidth: each thread expands 4 packed bytes
// (8 weights) per iteration, so reads are dwordx4-coalesced and writes are
// bf16x8 (16 B) stores.
//
// Codebook: the 16 NF4 quantiles of N(0,1) from the QLoRA paper (public).

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

__constant__ float NF4_CODE[16] = {
    -1.0f,
    -0.6961928009986877f,
    -0.5250730514526367f,
    -0.39491748809814453f,
    -0.28444138169288635f,
    -0.18477343022823334f,
    -0.09105003625154495f,
    0.0f,
    0.07958029955625534f,
    0.16093020141124725f,
    0.24611230194568634f,
    0.33791524171829224f,
    0.44070982933044434f,
    0.5626170039176941f,
    0.7229568362236023f,
    1.0f,
};

// block_size is the quantization block (64 weights = 32 packed bytes), so one
// thread-iteration of 8 weights never straddles two absmax blocks.
__global__ void nf4_dequant_kernel(const uint8_t* __restrict__ packed,
                                   const float* __restrict__ absmax,
                                   bf16* __restrict__ out, long n_half,
                                   int block_half) {
  // n_half = number of packed bytes (= numel/2); block_half = block_size/2.
  long i0 = (long)(blockIdx.x) * blockDim.x * 4 + threadIdx.x * 4;
  if (i0 >= n_half) return;
  // 4 bytes per thread, coalesced as one dword.
  uint32_t word;
  if (i0 + 4 <= n_half) {
    word = *reinterpret_cast<const uint32_t*>(packed + i0);
  } else {
    word = 0;
    for (long j = 0; i0 + j < n_half; ++j) word |= (uint32_t)packed[i0 + j] << (8 * j);
  }
  const float scale = absmax[i0 / block_half];
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const uint32_t byte = (word >> (8 * j)) & 0xffu;
    // low nibble = first (even-index) weight, high nibble = second.
    o.v[2 * j] = f2bf(NF4_CODE[byte & 0xf] * scale);
    o.v[2 * j + 1] = f2bf(NF4_CODE[byte >> 4] * scale);
  }
  if (i0 + 4 <= n_half) {
    *reinterpret_cast<bf16x8*>(out + 2 * i0) = o;
  } else {
    for (long j = 0; 2 * i0 + j < 2 * n_half; ++j) out[2 * i0 + j] = o.v[j];
  }
}

at::Tensor nf4_dequant(const at::Tensor& packed, const at::Tensor& absmax,
                          int64_t block_size, int64_t rows, int64_t cols) {
  TORCH_CHECK(packed.is_cuda() && packed.dtype() == at::kByte,
              "nf4_dequant: packed must be uint8 on GPU");
  TORCH_CHECK(absmax.is_cuda() && absmax.dtype() == at::kFloat,
              "nf4_dequant: absmax must be fp32 on GPU");
  TORCH_CHECK(block_size % 8 == 0 && block_size >= 8, "block_size multiple of 8");
  const long n_half = packed.numel();
  TORCH_CHECK(rows * cols == 2 * n_half, "shape mismatch");
  auto out = at::empty({rows, cols},
                          packed.options().dtype(at::kBFloat16));
  const int threads = 256;
  const long grid = (n_half + (long)threads * 4 - 1) / ((long)threads * 4);
  auto stream = c10::hip::getCurrentHIPStream();
  nf4_dequant_kernel<<<dim3((unsigned)grid), dim3(threads), 0, stream>>>(
      packed.data_ptr<uint8_t>(), absmax.data_ptr<float>(),
      reinterpret_cast<bf16*>(out.data_ptr()), n_half, (int)(block_size / 2));
  HIP_CHECK_KERNEL();
  return out;
}

}  // namespace amd_ops
