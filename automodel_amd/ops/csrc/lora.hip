// Fused LoRA forward (CDNA4): delta = (x @ A^T) @ B^T * scale in ONE kernel
// — the [M, r] intermediate lives in registers/LDS and never touches HBM,
// and one launch replaces two skinny hipBLASLt GEMMs per adapter.
//
// Reference: nemo_automodel/components/_peft/lora_kernel.py:182
// (lora_forward_kernel, Triton). Backward stays the composite two-GEMM
// formulas in python (peft/lora.py) — the adapter matrices are tiny, so the
// backward GEMMs are launch-cheap relative to the fused forward's saving.
//
// Shapes: x [M, H] bf16, A [r, H], B [O, r], r in {16, 32, 64},
// H % 64 == 0, O % 64 == 0. Block: 256 threads (4 waves), 256 m-rows
// (64 per wave); phase 1 accumulates t = x A^T with A staged per-H-chunk,
// phase 2 stages t per wave in LDS and streams B in 64-col chunks.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

typedef __bf16 bf16x8l __attribute__((ext_vector_type(8)));
typedef float f32x4l __attribute__((ext_vector_type(4)));

// [row][64-col] LDS tiles, 128-B rows: same rotation as the grouped kernels
__device__ __forceinline__ int lra_off(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ (((row >> 1) & 7) << 4));
}

template <int R>
__global__ __launch_bounds__(256) void lora_fused_fwd_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ A, const bf16* __restrict__ B,
    bf16* __restrict__ out, long M, int H, int O, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* a_lds = smem;                    // [R][64] staged A chunk (R*128 B)
  char* t_lds = smem + R * 128;          // [256 m][R] bf16 t values
  char* b_lds = t_lds + 256 * R * 2;     // [64 o][R] staged B chunk

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;
  const int kq = lane >> 4;

  const long m0 = (long)blockIdx.x * 256;
  const int mw = wid * 64;               // wave's m-offset within the block

  // ---- phase 1: t[m, r] = x[m, :] @ A^T  (acc [64 m][R] per wave)
  f32x4l acc1[4][R / 16];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < R / 16; ++j)
#pragma unroll
      for (int v = 0; v < 4; ++v) acc1[i][j][v] = 0.f;

  for (int h0 = 0; h0 < H; h0 += 64) {
    // stage A chunk [R][64] (vector)
    for (int idx = tid; idx < R * 8; idx += 256) {
      const int r = idx / 8, c = (idx % 8) * 8;
      *reinterpret_cast<bf16x8*>(a_lds + lra_off(r, c * 2)) =
          *reinterpret_cast<const bf16x8*>(A + (long)r * H + h0 + c);
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8l b[R / 16];
#pragma unroll
      for (int j = 0; j < R / 16; ++j)
        b[j] = *reinterpret_cast<const bf16x8l*>(
            a_lds + lra_off(j * 16 + l16, (kk * 32 + kq * 8) * 2));
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const long m = m0 + mw + i * 16 + l16;
        bf16x8l a;
        if (m < M) {
          a = *reinterpret_cast<const bf16x8l*>(x + m * H + h0 + kk * 32 + kq * 8);
        } else {
#pragma unroll
          for (int v = 0; v < 8; ++v) a[v] = (__bf16)0.f;
        }
#pragma unroll
        for (int j = 0; j < R / 16; ++j)
          acc1[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b[j], acc1[i][j], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- park t in LDS as [m][r] bf16 (D layout: col=l16, row=kq*4+v per tile)
  {
    char* tw = t_lds + (long)(mw) * R * 2;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < R / 16; ++j)
#pragma unroll
        for (int v = 0; v < 4; ++v) {
          const int m = i * 16 + kq * 4 + v;       // row within wave tile
          const int r = j * 16 + l16;
          *reinterpret_cast<bf16*>(tw + (long)m * R * 2 + r * 2) =
              f2bf(acc1[i][j][v]);
        }
  }
  __syncthreads();

  // ---- phase 2: out[m, o] = t[m, :] @ B^T * scale, streaming 64-o chunks
  for (int o0 = 0; o0 < O; o0 += 64) {
    for (int idx = tid; idx < 64 * (R / 8); idx += 256) {
      const int o = idx / (R / 8), c = (idx % (R / 8)) * 8;
      *reinterpret_cast<bf16x8*>(b_lds + lra_off(o, c * 2)) =
          *reinterpret_cast<const bf16x8*>(B + (long)(o0 + o) * R + c);
    }
    __syncthreads();

    f32x4l acc2[4][4];        // [m-tile][o-tile of 16]
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int v = 0; v < 4; ++v) acc2[i][j][v] = 0.f;

    const char* tw = t_lds + (long)(mw) * R * 2;
#pragma unroll
    for (int kk = 0; kk < R / 32; ++kk) {
      bf16x8l bfr[4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bfr[j] = *reinterpret_cast<const bf16x8l*>(
            b_lds + lra_off(j * 16 + l16, (kk * 32 + kq * 8) * 2));
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        bf16x8l a = *reinterpret_cast<const bf16x8l*>(
            tw + (long)(i * 16 + l16) * R * 2 + (kk * 32 + kq * 8) * 2);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc2[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[j], acc2[i][j], 0, 0, 0);
      }
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const long m = m0 + mw + i * 16 + kq * 4 + v;
        if (m < M) {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            out[m * O + o0 + j * 16 + l16] = f2bf(acc2[i][j][v] * scale);
        }
      }
    }
    __syncthreads();
  }
}

// ===========================================================================
// SGMV: segmented multi-adapter fused LoRA (serving path — SURVEY §2.9
// "SGMV kernels"; reference uses punica-style segmented gather MV for
// per-request adapters). Tokens arrive SORTED by adapter; the grouped-GEMM
// plan (offs + 256-row tile_map, device-built) picks each tile's adapter,
// whose A/B/scale are indexed from stacked tensors. Same two-phase
// register/LDS structure as the single-adapter kernel above.
// ===========================================================================

template <int R>
__global__ __launch_bounds__(256) void sgmv_fused_fwd_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ A, const bf16* __restrict__ B,
    const float* __restrict__ scales, bf16* __restrict__ out,
    const int* __restrict__ tile_map, const int* __restrict__ offs,
    const int* __restrict__ n_tiles, int H, int O) {
  if (n_tiles != nullptr && (int)blockIdx.x >= n_tiles[0]) return;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* a_lds = smem;
  char* t_lds = smem + R * 128;
  char* b_lds = t_lds + 256 * R * 2;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;
  const int kq = lane >> 4;

  const int ad = tile_map[2 * blockIdx.x];
  const long m0 = tile_map[2 * blockIdx.x + 1];
  const long m_end = offs[ad + 1];
  const float scale = scales[ad];
  const bf16* Aa = A + (long)ad * R * H;
  const bf16* Ba = B + (long)ad * O * R;
  const int mw = wid * 64;

  f32x4l acc1[4][R / 16];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < R / 16; ++j)
#pragma unroll
      for (int v = 0; v < 4; ++v) acc1[i][j][v] = 0.f;

  for (int h0 = 0; h0 < H; h0 += 64) {
    for (int idx = tid; idx < R * 8; idx += 256) {
      const int r = idx / 8, c = (idx % 8) * 8;
      *reinterpret_cast<bf16x8*>(a_lds + lra_off(r, c * 2)) =
          *reinterpret_cast<const bf16x8*>(Aa + (long)r * H + h0 + c);
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8l b[R / 16];
#pragma unroll
      for (int j = 0; j < R / 16; ++j)
        b[j] = *reinterpret_cast<const bf16x8l*>(
            a_lds + lra_off(j * 16 + l16, (kk * 32 + kq * 8) * 2));
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const long m = m0 + mw + i * 16 + l16;
        bf16x8l a;
        if (m < m_end) {
          a = *reinterpret_cast<const bf16x8l*>(x + m * H + h0 + kk * 32 + kq * 8);
        } else {
#pragma unroll
          for (int v = 0; v < 8; ++v) a[v] = (__bf16)0.f;
        }
#pragma unroll
        for (int j = 0; j < R / 16; ++j)
          acc1[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b[j], acc1[i][j], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  {
    char* tw = t_lds + (long)(mw) * R * 2;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < R / 16; ++j)
#pragma unroll
        for (int v = 0; v < 4; ++v) {
          const int m = i * 16 + kq * 4 + v;
          const int r = j * 16 + l16;
          *reinterpret_cast<bf16*>(tw + (long)m * R * 2 + r * 2) =
              f2bf(acc1[i][j][v]);
        }
  }
  __syncthreads();

  for (int o0 = 0; o0 < O; o0 += 64) {
    for (int idx = tid; idx < 64 * (R / 8); idx += 256) {
      const int o = idx / (R / 8), c = (idx % (R / 8)) * 8;
      *reinterpret_cast<bf16x8*>(b_lds + lra_off(o, c * 2)) =
          *reinterpret_cast<const bf16x8*>(Ba + (long)(o0 + o) * R + c);
    }
    __syncthreads();

    f32x4l acc2[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int v = 0; v < 4; ++v) acc2[i][j][v] = 0.f;

    const char* tw = t_lds + (long)(mw) * R * 2;
#pragma unroll
    for (int kk = 0; kk < R / 32; ++kk) {
      bf16x8l bfr[4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bfr[j] = *reinterpret_cast<const bf16x8l*>(
            b_lds + lra_off(j * 16 + l16, (kk * 32 + kq * 8) * 2));
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        bf16x8l a = *reinterpret_cast<const bf16x8l*>(
            tw + (long)(i * 16 + l16) * R * 2 + (kk * 32 + kq * 8) * 2);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc2[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[j], acc2[i][j], 0, 0, 0);
      }
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const long m = m0 + mw + i * 16 + kq * 4 + v;
        if (m < m_end) {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            out[m * O + o0 + j * 16 + l16] = f2bf(acc2[i][j][v] * scale);
        }
      }
    }
    __syncthreads();
  }
}

at::Tensor sgmv_fused_fwd(const at::Tensor& x, const at::Tensor& A, const at::Tensor& B,
                          const at::Tensor& scales, const at::Tensor& offs,
                          const at::Tensor& tile_map,
                          const std::optional<at::Tensor>& n_tiles) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.scalar_type() == at::kBFloat16,
              "sgmv_fused_fwd: x [M,H] bf16");
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3, "A [n,r,H], B [n,O,r]");
  const long M = x.size(0);
  const int H = x.size(1), R = A.size(1), O = B.size(1);
  TORCH_CHECK(A.size(2) == H && B.size(2) == R && A.size(0) == B.size(0),
              "shape mismatch");
  TORCH_CHECK((R == 32 || R == 64) && H % 64 == 0 && O % 64 == 0,
              "sgmv: r in {32, 64}, H%64==0, O%64==0");
  TORCH_CHECK(scales.scalar_type() == at::kFloat && scales.numel() == A.size(0),
              "scales: float32 [n_adapters]");
  auto out = at::empty({M, (long)O}, x.options());
  const int nt = tile_map.size(0);
  if (M == 0 || nt == 0) return out;
  const size_t smem = R * 128 + 256 * R * 2 + 64 * 128;
  auto stream = c10::hip::getCurrentHIPStream();
  const int* ntp = n_tiles.has_value() ? n_tiles->data_ptr<int>() : nullptr;
  if (R == 32) {
    hipLaunchKernelGGL((sgmv_fused_fwd_kernel<32>), dim3(nt), dim3(256), smem,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<const bf16*>(A.data_ptr()),
                       reinterpret_cast<const bf16*>(B.data_ptr()),
                       scales.data_ptr<float>(),
                       reinterpret_cast<bf16*>(out.data_ptr()),
                       tile_map.data_ptr<int>(), offs.data_ptr<int>(), ntp, H, O);
  } else {
    hipLaunchKernelGGL((sgmv_fused_fwd_kernel<64>), dim3(nt), dim3(256), smem,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<const bf16*>(A.data_ptr()),
                       reinterpret_cast<const bf16*>(B.data_ptr()),
                       scales.data_ptr<float>(),
                       reinterpret_cast<bf16*>(out.data_ptr()),
                       tile_map.data_ptr<int>(), offs.data_ptr<int>(), ntp, H, O);
  }
  HIP_CHECK_KERNEL();
  return out;
}

at::Tensor lora_fused_fwd(const at::Tensor& x, const at::Tensor& A, const at::Tensor& B,
                          double scale) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.scalar_type() == at::kBFloat16,
              "lora_fused_fwd: x [M,H] bf16");
  const long M = x.size(0);
  const int H = x.size(1), R = A.size(0), O = B.size(0);
  TORCH_CHECK(A.size(1) == H && B.size(1) == R, "shape mismatch");
  TORCH_CHECK((R == 32 || R == 64) && H % 64 == 0 && O % 64 == 0,
              "lora_fused_fwd: r in {32, 64}, H%64==0, O%64==0");
  auto out = at::empty({M, (long)O}, x.options());
  if (M == 0) return out;
  const long grid = (M + 255) / 256;
  const size_t smem = R * 128 + 256 * R * 2 + 64 * 128;
  auto stream = c10::hip::getCurrentHIPStream();
  if (R == 32) {
    hipLaunchKernelGGL((lora_fused_fwd_kernel<32>), dim3((unsigned)grid), dim3(256),
                       smem, stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<const bf16*>(A.data_ptr()),
                       reinterpret_cast<const bf16*>(B.data_ptr()),
                       reinterpret_cast<bf16*>(out.data_ptr()), M, H, O, (float)scale);
  } else {
    hipLaunchKernelGGL((lora_fused_fwd_kernel<64>), dim3((unsigned)grid), dim3(256),
                       smem, stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<const bf16*>(A.data_ptr()),
                       reinterpret_cast<const bf16*>(B.data_ptr()),
                       reinterpret_cast<bf16*>(out.data_ptr()), M, H, O, (float)scale);
  }
  HIP_CHECK_KERNEL();
  return out;
}

}  // namespace amd_ops
