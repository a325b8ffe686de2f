// Shared helpers for the automodel_amd CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA4 — every cross-lane idiom below assumes it
// (guide: cdna_hip_programming.md §1).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64

using bf16 = __hip_bfloat16;

// 16-byte vector of 8 bf16 — loads/stores compile to global_load_dwordx4
// (guide Guideline 13: ALWAYS vectorize bf16 loads).
struct alignas(16) bf16x8 {
  bf16 v[8];
};
struct alignas(8) bf16x4 {
  bf16 v[4];
};

__device__ __forceinline__ float bf2f(bf16 x) { return __bfloat162float(x); }
__device__ __forceinline__ bf16 f2bf(float x) { return __float2bfloat16(x); }

// Full-wave (64-lane) sum reduction.
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off));
  return x;
}

// Block-level sum reduction; `scratch` must hold >= blockDim.x/64 floats.
// Returns the total to every thread.
__device__ __forceinline__ float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

#define HIP_CHECK_KERNEL()                                              \
  do {                                                                  \
    hipError_t err_ = hipGetLastError();                                \
    TORCH_CHECK(err_ == hipSuccess, "HIP kernel launch failed: ",       \
                hipGetErrorString(err_));                               \
  } while (0)

__host__ __forceinline__ int ceil_div_i(long a, long b) { return (int)((a + b - 1) / b); }
