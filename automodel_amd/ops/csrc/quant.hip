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


// ===========================================================================
// FP8 training casts (round 2): single-pass bf16 -> e4m3fn/e5m2 with the
// tensor amax recorded as a side effect (delayed scaling: the NEXT step's
// scale comes from this step's amax — reference torchao delayed recipe).
// The round-1 torch-chain cast (float() upcast + amax pass + clamp + cast)
// cost 0.85 ms per [32768,4096] operand (benchmarks/fp8_micro.py); this is
// one read + one 1-byte write + a wave-reduced atomic max (~0.1 ms).
// gfx950 packs via cvt_pk_fp8_f32 (OCP e4m3fn) / cvt_pk_bf8_f32 (e5m2).
// ===========================================================================

__global__ void fp8_cast_kernel(const bf16* __restrict__ x, unsigned char* __restrict__ out,
                                const float* __restrict__ scale,
                                float* __restrict__ amax_out, long n8,
                                float maxv, bool e5m2) {
  const float s = scale[0];
  float local_max = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 v = reinterpret_cast<const bf16x8*>(x)[i];
    float f[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float raw = bf2f(v.v[j]);
      local_max = fmaxf(local_max, fabsf(raw));
      f[j] = fminf(fmaxf(raw * s, -maxv), maxv);
    }
    unsigned d0 = 0, d1 = 0;
    if (e5m2) {
      d0 = __builtin_amdgcn_cvt_pk_bf8_f32(f[0], f[1], d0, false);
      d0 = __builtin_amdgcn_cvt_pk_bf8_f32(f[2], f[3], d0, true);
      d1 = __builtin_amdgcn_cvt_pk_bf8_f32(f[4], f[5], d1, false);
      d1 = __builtin_amdgcn_cvt_pk_bf8_f32(f[6], f[7], d1, true);
    } else {
      d0 = __builtin_amdgcn_cvt_pk_fp8_f32(f[0], f[1], d0, false);
      d0 = __builtin_amdgcn_cvt_pk_fp8_f32(f[2], f[3], d0, true);
      d1 = __builtin_amdgcn_cvt_pk_fp8_f32(f[4], f[5], d1, false);
      d1 = __builtin_amdgcn_cvt_pk_fp8_f32(f[6], f[7], d1, true);
    }
    uint2 packed{d0, d1};
    reinterpret_cast<uint2*>(out)[i] = packed;
  }
  // tensor amax: wave reduce + atomic max (positive floats compare as ints)
  local_max = wave_reduce_max(local_max);
  if ((threadIdx.x & 63) == 0)
    atomicMax(reinterpret_cast<int*>(amax_out), __float_as_int(local_max));
}

// byte-level tiled transpose for the fp8 wgrad operands ([R,C] -> [C,R])
__global__ __launch_bounds__(256) void fp8_transpose_kernel(
    const unsigned char* __restrict__ in, unsigned char* __restrict__ out,
    int R, int C) {
  __shared__ unsigned char tile[64][64 + 8];
  const int tr0 = blockIdx.y * 64, tc0 = blockIdx.x * 64;
  const int tid = threadIdx.x;
  for (int idx = tid; idx < 64 * 8; idx += 256) {
    const int r = idx / 8, c = (idx % 8) * 8;
    const int gr = tr0 + r, gc = tc0 + c;
    if (gr < R && gc + 7 < C) {
      const unsigned long long v =
          *reinterpret_cast<const unsigned long long*>(in + (long)gr * C + gc);
#pragma unroll
      for (int j = 0; j < 8; ++j) tile[c + j][r] = (unsigned char)(v >> (8 * j));
    } else if (gr < R) {
      for (int j = 0; j < 8 && gc + j < C; ++j) tile[c + j][r] = in[(long)gr * C + gc + j];
    }
  }
  __syncthreads();
  for (int idx = tid; idx < 64 * 8; idx += 256) {
    const int r = idx / 8, c = (idx % 8) * 8;
    const int gr = tc0 + r, gc = tr0 + c;
    if (gr < C && gc + 7 < R) {
      unsigned long long v = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j) v |= (unsigned long long)tile[r][c + j] << (8 * j);
      *reinterpret_cast<unsigned long long*>(out + (long)gr * R + gc) = v;
    } else if (gr < C) {
      for (int j = 0; j < 8 && gc + j < R; ++j) out[(long)gr * R + gc + j] = tile[r][c + j];
    }
  }
}

at::Tensor fp8_cast(const at::Tensor& x, const at::Tensor& scale, at::Tensor amax_out,
                    bool e5m2) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous(),
              "fp8_cast: contiguous bf16");
  TORCH_CHECK(x.numel() % 8 == 0, "fp8_cast: numel % 8 == 0");
  auto dt = e5m2 ? at::kFloat8_e5m2 : at::kFloat8_e4m3fn;
  auto out = at::empty_like(x, x.options().dtype(dt));
  const long n8 = x.numel() / 8;
  const int block = 256;
  const int grid = (int)std::min<long>((n8 + block - 1) / block, 4096);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fp8_cast_kernel, dim3(grid), dim3(block), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(x.data_ptr()),
                     reinterpret_cast<unsigned char*>(out.data_ptr()),
                     scale.data_ptr<float>(), amax_out.data_ptr<float>(), n8,
                     e5m2 ? 57344.f : 448.f, e5m2);
  HIP_CHECK_KERNEL();
  return out;
}

at::Tensor fp8_transpose(const at::Tensor& x8) {
  TORCH_CHECK(x8.is_cuda() && x8.dim() == 2 && x8.element_size() == 1 &&
              x8.is_contiguous(), "fp8_transpose: contiguous 2-D fp8");
  const int R = x8.size(0), C = x8.size(1);
  auto out = at::empty({C, R}, x8.options());
  const dim3 grid((C + 63) / 64, (R + 63) / 64);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fp8_transpose_kernel, grid, dim3(256), 0, stream.stream(),
                     reinterpret_cast<const unsigned char*>(x8.data_ptr()),
                     reinterpret_cast<unsigned char*>(out.data_ptr()), R, C);
  HIP_CHECK_KERNEL();
  return out;
}

}  // namespace amd_ops
