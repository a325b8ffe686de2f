// Fused linear cross-entropy forward (lm_head GEMM + softmax stats, no
// logits tensor). MI355X-native cut-cross-entropy equivalent
// (reference: nemo_automodel/components/loss/linear_ce.py:253).
//
// loss_i = lse_i - <h_i, w_{y_i}>,  lse_i = logsumexp_v <h_i, w_v>
//
// Three deterministic kernels (no atomics):
//   1. ce_stats GEMM: grid (T/128, V/256); each 8-wave block computes its
//      [128 token x 256 vocab] S-tile with mfma_32x32x16_bf16 in the SWAPPED
//      layout (A = W rows, B = H rows -> each lane holds S columns for ONE
//      token row, so max/sum reduce in-register + one __shfl_xor(32)), then
//      writes per-tile (m, l) partials [nV][T] — 2*4 B per tile-row instead
//      of 512 B of logits.
//   2. label-dot: logit_y[i] = <h_i, W[y_i]> (one wave per token).
//   3. combine: lse[i] = m + log(sum_j exp(m_j - m) l_j); per-token loss
//      written to a [T] buffer (summed by torch — deterministic reduction).
//
// Backward stays the hybrid chunked-recompute path (hipBLASLt GEMM + the
// in-place CE epilogue of ce_logits.hip): dW/dH need a [chunk, V] k-split
// workspace in ANY scheme, and hipBLASLt runs the three big GEMMs at peak —
// a fused backward would re-derive the same traffic with slower MFMA code.
// The round-1 stub and its silent fallback are gone.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

typedef __bf16 bf16x8_v __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

// [row][64 B] LDS images (KT=32 bf16 per row): same conflict-free rotation
// as the flash kernels' transposed images.
__device__ __forceinline__ int ce_lds_off(int row, int byte_in_row) {
  return row * 64 + (byte_in_row ^ ((((row >> 2) ^ (row >> 3)) & 3) << 4));
}

#define CE_TM 128   // token rows per block
#define CE_TN 256   // vocab cols per block
#define CE_KT 32    // k (hidden) chunk

// 8 waves as 4 (token groups of 32) x 2 (vocab groups of 128).
__global__ __launch_bounds__(512, 2) void ce_stats_kernel(
    const bf16* __restrict__ hidden, const bf16* __restrict__ weight,
    float* __restrict__ part_m, float* __restrict__ part_l,
    int T, int V, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* h_lds = smem;                    // [CE_TM][64 B]  = 8 KiB
  char* w_lds = smem + CE_TM * 64;       // [CE_TN][64 B]  = 16 KiB
  float* red = reinterpret_cast<float*>(smem + (CE_TM + CE_TN) * 64);  // [2][4][32]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;           // token row within wave tile
  const int half = lane >> 5;
  const int wm = wid & 3;              // token group
  const int wn = wid >> 2;             // vocab group

  const int t0 = blockIdx.x * CE_TM;   // block token rows [t0, t0+128)
  const int v0 = blockIdx.y * CE_TN;   // block vocab cols [v0, v0+256)

  f32x16 acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc[t][r] = 0.f;

  const long h_rs = H, w_rs = H;
  const int n_k = H / CE_KT;           // H % 32 == 0 (checked host-side)

  // T14 split: issue global loads for chunk kc while computing kc-1.
  // Each thread stages: 1 H vector (128 rows x 4 chunks = 512 threads) and
  // 2 W vectors (256 rows x 4 chunks = 1024).
  const int h_row = tid >> 2, h_ch = (tid & 3) * 8;
  const int w_row0 = tid >> 1, w_ch0 = (tid & 1) * 16;  // two vectors: ch0, ch0+8
  bf16x8 h_reg, w_reg0, w_reg1;
  const bool h_live = t0 + h_row < T;
  const bool w_live = v0 + w_row0 < V;

  auto issue_loads = [&](int kc) {
    const int kb = kc * CE_KT;
    h_reg = h_live ? *reinterpret_cast<const bf16x8*>(
                         hidden + (long)(t0 + h_row) * h_rs + kb + h_ch)
                   : bf16x8{};
    if (w_live) {
      w_reg0 = *reinterpret_cast<const bf16x8*>(
          weight + (long)(v0 + w_row0) * w_rs + kb + w_ch0);
      w_reg1 = *reinterpret_cast<const bf16x8*>(
          weight + (long)(v0 + w_row0) * w_rs + kb + w_ch0 + 8);
    } else {
      w_reg0 = bf16x8{};
      w_reg1 = bf16x8{};
    }
  };
  auto write_lds = [&] {
    *reinterpret_cast<bf16x8*>(h_lds + ce_lds_off(h_row, h_ch * 2)) = h_reg;
    *reinterpret_cast<bf16x8*>(w_lds + ce_lds_off(w_row0, w_ch0 * 2)) = w_reg0;
    *reinterpret_cast<bf16x8*>(w_lds + ce_lds_off(w_row0, (w_ch0 + 8) * 2)) = w_reg1;
  };

  issue_loads(0);
  for (int kc = 0; kc < n_k; ++kc) {
    write_lds();
    __syncthreads();
    if (kc + 1 < n_k) issue_loads(kc + 1);
    // compute chunk kc from LDS
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {          // two k=16 halves of KT=32
      bf16x8_v hb = *reinterpret_cast<const bf16x8_v*>(
          h_lds + ce_lds_off(wm * 32 + col, (kh * 16 + half * 8) * 2));
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        bf16x8_v wa = *reinterpret_cast<const bf16x8_v*>(
            w_lds + ce_lds_off(wn * 128 + t * 32 + col, (kh * 16 + half * 8) * 2));
        acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(wa, hb, acc[t], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- per-lane online stats over the wave's 128 vocab cols
  float m = -1e30f, l = 0.f;
#pragma unroll
  for (int t = 0; t < 4; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int vg = v0 + wn * 128 + t * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (vg < V) m = fmaxf(m, acc[t][r]);
    }
  }
  m = fmaxf(m, __shfl_xor(m, 32));
#pragma unroll
  for (int t = 0; t < 4; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int vg = v0 + wn * 128 + t * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (vg < V) l += __expf(acc[t][r] - m);
    }
  }
  l += __shfl_xor(l, 32);

  // ---- combine the two vocab groups (wn 0/1) per token row via LDS
  if (lane < 32) {
    red[(wn * 4 + wm) * 32 + lane] = m;
    red[(8 + wn * 4 + wm) * 32 + lane] = l;
  }
  __syncthreads();
  if (wn == 0 && lane < 32) {
    const int tok = t0 + wm * 32 + lane;
    if (tok < T) {
      const float m0 = red[(0 + wm) * 32 + lane], m1 = red[(4 + wm) * 32 + lane];
      const float l0 = red[(8 + wm) * 32 + lane], l1 = red[(12 + wm) * 32 + lane];
      const float mm = fmaxf(m0, m1);
      const float ll = __expf(m0 - mm) * l0 + __expf(m1 - mm) * l1;
      const long pidx = (long)blockIdx.y * T + tok;
      part_m[pidx] = mm;
      part_l[pidx] = ll;
    }
  }
}

// one wave per token: logit_y[i] = <h_i, W[y_i]>
__global__ void ce_label_dot_kernel(const bf16* __restrict__ hidden,
                                    const bf16* __restrict__ weight,
                                    const long* __restrict__ labels,
                                    float* __restrict__ logit_y, int T, int H,
                                    long ignore_index) {
  const long t = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE_SIZE;
  if (t >= T) return;
  const int lane = threadIdx.x & 63;
  const long y = labels[t];
  if (y == ignore_index) {
    if (lane == 0) logit_y[t] = 0.f;
    return;
  }
  const bf16* h = hidden + t * H;
  const bf16* w = weight + y * H;
  float acc = 0.f;
  for (int j = lane * 8; j < H; j += 64 * 8) {
    bf16x8 hv = *reinterpret_cast<const bf16x8*>(h + j);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + j);
#pragma unroll
    for (int u = 0; u < 8; ++u) acc += bf2f(hv.v[u]) * bf2f(wv.v[u]);
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) logit_y[t] = acc;
}

// lse[i] from the [nV][T] partials; per-token loss (0 for ignored labels)
__global__ void ce_combine_kernel(const float* __restrict__ part_m,
                                  const float* __restrict__ part_l,
                                  const float* __restrict__ logit_y,
                                  const long* __restrict__ labels,
                                  float* __restrict__ lse, float* __restrict__ loss,
                                  int T, int nV, long ignore_index) {
  const long t = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= T) return;
  float m = -1e30f;
  for (int j = 0; j < nV; ++j) m = fmaxf(m, part_m[(long)j * T + t]);
  float l = 0.f;
  for (int j = 0; j < nV; ++j)
    l += __expf(part_m[(long)j * T + t] - m) * part_l[(long)j * T + t];
  const float ls = m + __logf(l);
  lse[t] = ls;
  loss[t] = (labels[t] == ignore_index) ? 0.f : (ls - logit_y[t]);
}

std::tuple<at::Tensor, at::Tensor> fused_ce_fwd(const at::Tensor& hidden,
                                                const at::Tensor& weight,
                                                const at::Tensor& labels) {
  TORCH_CHECK(hidden.is_cuda() && hidden.dim() == 2 &&
                  hidden.scalar_type() == at::kBFloat16,
              "fused_ce_fwd: hidden must be [T,H] bf16 on GPU");
  TORCH_CHECK(weight.dim() == 2 && weight.scalar_type() == at::kBFloat16,
              "fused_ce_fwd: weight must be [V,H] bf16");
  TORCH_CHECK(labels.scalar_type() == at::kLong, "labels must be int64");
  const int T = hidden.size(0), H = hidden.size(1), V = weight.size(0);
  TORCH_CHECK(H % CE_KT == 0, "fused_ce_fwd: H must be a multiple of 32");
  TORCH_CHECK(hidden.is_contiguous() && weight.is_contiguous());
  auto stream = c10::hip::getCurrentHIPStream().stream();

  const int nT = (T + CE_TM - 1) / CE_TM;
  const int nV = (V + CE_TN - 1) / CE_TN;
  auto opts = hidden.options().dtype(at::kFloat);
  auto part_m = at::empty({nV, T}, opts);
  auto part_l = at::empty({nV, T}, opts);
  auto logit_y = at::empty({T}, opts);
  auto lse = at::empty({T}, opts);
  auto loss = at::empty({T}, opts);

  {
    const dim3 grid(nT, nV);
    const size_t smem = (CE_TM + CE_TN) * 64 + 16 * 32 * sizeof(float);
    hipLaunchKernelGGL(ce_stats_kernel, grid, dim3(512), smem, stream,
                       reinterpret_cast<const bf16*>(hidden.data_ptr()),
                       reinterpret_cast<const bf16*>(weight.data_ptr()),
                       part_m.data_ptr<float>(), part_l.data_ptr<float>(), T, V, H);
    HIP_CHECK_KERNEL();
  }
  {
    const int block = 256;
    const long grid = ((long)T * WAVE_SIZE + block - 1) / block;
    hipLaunchKernelGGL(ce_label_dot_kernel, dim3((unsigned)grid), dim3(block), 0, stream,
                       reinterpret_cast<const bf16*>(hidden.data_ptr()),
                       reinterpret_cast<const bf16*>(weight.data_ptr()),
                       labels.data_ptr<long>(), logit_y.data_ptr<float>(), T, H, -100);
    HIP_CHECK_KERNEL();
  }
  {
    const int block = 256;
    hipLaunchKernelGGL(ce_combine_kernel, dim3((T + block - 1) / block), dim3(block), 0,
                       stream, part_m.data_ptr<float>(), part_l.data_ptr<float>(),
                       logit_y.data_ptr<float>(), labels.data_ptr<long>(),
                       lse.data_ptr<float>(), loss.data_ptr<float>(), T, nV, -100);
    HIP_CHECK_KERNEL();
  }
  // torch sum = deterministic tree reduction (no atomics anywhere above)
  return {loss.sum(), lse};
}

}  // namespace amd_ops
