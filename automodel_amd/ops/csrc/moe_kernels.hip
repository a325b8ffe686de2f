// MoE kernels (CDNA4): grouped GEMM (bf16, MFMA 16x16x32) + fused token
// permute/unpermute.
//
// Replaces the reference's external grouped_gemm / torch._grouped_mm expert
// path and the fused permute/unpermute utilities (#5/#7):
//   grouped_gemm_nt: y[m, n] = sum_k x[m, k] * w[e(m), n, k] over variable-
//     size expert groups; one 128x128x64 LDS-tiled MFMA block per output
//     tile, tile list precomputed host-side (expert, m0).
//   permute_gather / unpermute_combine: gather token replicas into expert-
//     sorted order and combine top-k outputs with routing probs without
//     materializing the repeat_interleave copy.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

typedef __bf16 bf16x8v __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define GG_BM 128
#define GG_BN 128
#define GG_BK 64

// LDS tiles: [128 rows][64 cols] bf16, 128-B rows (32 words): rows of equal
// parity share a 32-word bank window; rotating the 16-B slot by (row>>1)&7
// separates the 8 colliding rows of a 16-lane b128 group -> conflict-free.
__device__ __forceinline__ int gg_off(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ (((row >> 1) & 7) << 4));
}

// BK-parametrized row offset: 128-B rows keep the (row>>1)&7 rotation;
// 256-B rows start at bank 0 every row, so rotate by row&15 (Guideline 4).
template <int BK>
__device__ __forceinline__ int ggk_off(int row, int byte_in_row) {
  if constexpr (BK == 64) {
    return row * 128 + (byte_in_row ^ (((row >> 1) & 7) << 4));
  } else {
    return row * (BK * 2) + (byte_in_row ^ ((row & 15) << 4));
  }
}

// Device-side group plan: offsets + (expert, row0) tile list built from the
// on-device counts so the hot path never syncs counts to the host
// (the round-1 tile map was host-built from counts.tolist() —
// one device sync per MoE layer per direction). One block, serial scan over
// E <= 4096 groups (microseconds).
__global__ void build_group_plan_kernel(const int* __restrict__ counts, int E,
                                        int* __restrict__ offs,
                                        int* __restrict__ tile_map,
                                        int* __restrict__ n_tiles, int bm) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  int acc = 0, t = 0;
  offs[0] = 0;
  for (int e = 0; e < E; ++e) {
    const int c = counts[e];
    for (int r = 0; r < c; r += bm) {
      tile_map[2 * t] = e;
      tile_map[2 * t + 1] = acc + r;
      ++t;
    }
    acc += c;
    offs[e + 1] = acc;
  }
  n_tiles[0] = t;
}

template <int BK>
__global__ __launch_bounds__(256) void grouped_gemm_nt_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w, bf16* __restrict__ y,
    const int* __restrict__ tile_map,   // [n_mtiles][2]: (expert, row0)
    const int* __restrict__ offs,       // [E+1] group row offsets
    const int* __restrict__ n_tiles,    // device tile count (null = grid-sized)
    int K, int N, int nfirst) {
  const int tb = nfirst ? (int)blockIdx.y : (int)blockIdx.x;
  const int nb = nfirst ? (int)blockIdx.x : (int)blockIdx.y;
  if (n_tiles != nullptr && tb >= n_tiles[0]) return;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* xa = smem;                  // [128][BK] bf16
  char* wb = smem + GG_BM * BK * 2;

  const int e = tile_map[2 * tb];
  const int m0 = tile_map[2 * tb + 1];
  const int m_end = offs[e + 1];
  const int n0 = nb * GG_BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;
  const int kq = lane >> 4;          // 0..3 -> k chunk of 8

  const long wbase = (long)e * N * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  constexpr int CPR = BK / 8;              // 16-B chunks per row
  constexpr int PASSES = 128 * CPR / 256;  // staging passes per tensor
  const int row0 = tid / CPR;
  const int c0 = (tid % CPR) * 8;

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- stage x tile [128 m][BK k] and w tile [128 n][BK k]
#pragma unroll
    for (int rr = 0; rr < PASSES; ++rr) {
      const int row = row0 + rr * (256 / CPR);
      const int m = m0 + row;
      bf16x8 xv;
      if (m < m_end) {
        xv = *reinterpret_cast<const bf16x8*>(x + (long)m * K + k0 + c0);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) xv.v[j] = f2bf(0.f);
      }
      *reinterpret_cast<bf16x8*>(xa + ggk_off<BK>(row, c0 * 2)) = xv;
      bf16x8 wv = *reinterpret_cast<const bf16x8*>(
          w + wbase + (long)(n0 + row) * K + k0 + c0);
      *reinterpret_cast<bf16x8*>(wb + ggk_off<BK>(row, c0 * 2)) = wv;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8v a[4], b[4];
      const int arow = (wid >> 1) * 64;
      const int brow = (wid & 1) * 64;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        a[i] = *reinterpret_cast<const bf16x8v*>(
            xa + ggk_off<BK>(arow + i * 16 + l16, (kk * 32 + kq * 8) * 2));
        b[i] = *reinterpret_cast<const bf16x8v*>(
            wb + ggk_off<BK>(brow + i * 16 + l16, (kk * 32 + kq * 8) * 2));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: D 16x16 layout col=l&15, row=(l>>4)*4+r
  const int mw = m0 + (wid >> 1) * 64;
  const int nw = n0 + (wid & 1) * 64;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mw + i * 16 + kq * 4 + r;
      if (m < m_end) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          y[(long)m * N + nw + j * 16 + l16] = f2bf(acc[i][j][r]);
        }
      }
    }
  }
}

// 256x256 big-tile NT variant (8 waves): 2x the arithmetic intensity per
// staged byte of the 128x128 tile — the 128-tile kernel parks 81% of wave
// cycles on memory waits (profiles/gg PMC) while hipBLASLt's MT256x256
// equivalents run 11%. Wave grid 4(m) x 2(n): 64 rows x 128 cols each.
__global__ __launch_bounds__(512, 1) void grouped_gemm_nt_big_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w, bf16* __restrict__ y,
    const int* __restrict__ tile_map, const int* __restrict__ offs,
    const int* __restrict__ n_tiles, int K, int N, int nfirst) {
  // nfirst: interpret blockIdx.x as the N tile (consecutive blocks then
  // share one (expert, m0) row tile — XCD round-robin dispatch puts the
  // sibling n-tiles of one x tile on different XCDs; A/B via
  // AMD_OPS_GG_NFIRST, benchmarks/gg_micro.py)
  const int tb = nfirst ? (int)blockIdx.y : (int)blockIdx.x;
  const int nb = nfirst ? (int)blockIdx.x : (int)blockIdx.y;
  if (n_tiles != nullptr && tb >= n_tiles[0]) return;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* xa = smem;                   // [256 m][64 k]  32 KiB
  char* wb = smem + 256 * 64 * 2;    // [256 n][64 k]  32 KiB

  const int e = tile_map[2 * tb];
  const int m0 = tile_map[2 * tb + 1];
  const int m_end = offs[e + 1];
  const int n0 = nb * 256;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;
  const int kq = lane >> 4;

  const long wbase = (long)e * N * K;

  f32x4 acc[4][8];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  const int row0 = tid / 8;          // 64 rows per pass (of 256)
  const int c0 = (tid % 8) * 8;

  for (int k0 = 0; k0 < K; k0 += 64) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int row = row0 + rr * 64;
      const int m = m0 + row;
      bf16x8 xv;
      if (m < m_end) {
        xv = *reinterpret_cast<const bf16x8*>(x + (long)m * K + k0 + c0);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) xv.v[j] = f2bf(0.f);
      }
      *reinterpret_cast<bf16x8*>(xa + gg_off(row, c0 * 2)) = xv;
      bf16x8 wv = *reinterpret_cast<const bf16x8*>(
          w + wbase + (long)(n0 + row) * K + k0 + c0);
      *reinterpret_cast<bf16x8*>(wb + gg_off(row, c0 * 2)) = wv;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8v a[4], b[8];
      const int arow = (wid >> 1) * 64;     // m group (4)
      const int brow = (wid & 1) * 128;     // n group (2)
#pragma unroll
      for (int i = 0; i < 4; ++i)
        a[i] = *reinterpret_cast<const bf16x8v*>(
            xa + gg_off(arow + i * 16 + l16, (kk * 32 + kq * 8) * 2));
#pragma unroll
      for (int j = 0; j < 8; ++j)
        b[j] = *reinterpret_cast<const bf16x8v*>(
            wb + gg_off(brow + j * 16 + l16, (kk * 32 + kq * 8) * 2));
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  const int mw = m0 + (wid >> 1) * 64;
  const int nw = n0 + (wid & 1) * 128;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mw + i * 16 + kq * 4 + r;
      if (m < m_end) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          y[(long)m * N + nw + j * 16 + l16] = f2bf(acc[i][j][r]);
      }
    }
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> build_group_plan(
    const at::Tensor& counts, int64_t M, int64_t bm) {
  TORCH_CHECK(counts.is_cuda() && counts.scalar_type() == at::kInt,
              "build_group_plan: counts int32 on GPU");
  const int E = counts.size(0);
  const long max_tiles = M / bm + E + 1;
  auto opts = counts.options();
  auto offs = at::empty({E + 1}, opts);
  auto tile_map = at::empty({max_tiles, 2}, opts);
  auto n_tiles = at::empty({1}, opts);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(build_group_plan_kernel, dim3(1), dim3(64), 0, stream.stream(),
                     counts.data_ptr<int>(), E, offs.data_ptr<int>(),
                     tile_map.data_ptr<int>(), n_tiles.data_ptr<int>(), (int)bm);
  HIP_CHECK_KERNEL();
  return {offs, tile_map, n_tiles};
}


static int gg_nfirst() {
  static const int v = []{
    const char* e = getenv("AMD_OPS_GG_NFIRST");
    return e ? atoi(e) : 1;
  }();
  return v;
}

at::Tensor grouped_gemm_nt(const at::Tensor& x, const at::Tensor& w,
                           const at::Tensor& offs, const at::Tensor& tile_map,
                           const std::optional<at::Tensor>& n_tiles, int64_t bm) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.scalar_type() == at::kBFloat16,
              "grouped_gemm_nt: x [M,K] bf16");
  TORCH_CHECK(w.dim() == 3 && w.scalar_type() == at::kBFloat16, "w [E,N,K] bf16");
  const long M = x.size(0);
  const int K = x.size(1), N = w.size(1);
  TORCH_CHECK(K == w.size(2), "K mismatch");
  TORCH_CHECK(K % GG_BK == 0 && N % GG_BN == 0, "need K%64==0, N%128==0");
  TORCH_CHECK(tile_map.scalar_type() == at::kInt && offs.scalar_type() == at::kInt,
              "tile_map/offs must be int32");
  auto y = at::empty({M, (long)N}, x.options());
  const int n_mtiles = tile_map.size(0);
  if (n_mtiles == 0 || M == 0) return y;
  auto stream = c10::hip::getCurrentHIPStream();
  const int* ntp = n_tiles.has_value() ? n_tiles->data_ptr<int>() : nullptr;
  if (bm == 256) {
    TORCH_CHECK(N % 256 == 0, "big-tile nt needs N%256==0");
    // n-first default: consecutive blocks share one (expert, m0) x-tile
    // and their sibling n-tiles round-robin across XCDs — measured +6-9%
    // (benchmarks/gg_grid_ab.py: 478->519 / 505->534 / 503->546 TF/s)
    const int nfirst = gg_nfirst();
    const dim3 gridb = nfirst ? dim3(N / 256, n_mtiles) : dim3(n_mtiles, N / 256);
    const size_t smemb = 2 * 256 * 64 * 2;
    hipLaunchKernelGGL(grouped_gemm_nt_big_kernel, gridb, dim3(512), smemb,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<const bf16*>(w.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       tile_map.data_ptr<int>(), offs.data_ptr<int>(), ntp, K, N,
                       nfirst);
    HIP_CHECK_KERNEL();
    return y;
  }
  const int nfirst = gg_nfirst();
  const dim3 grid = nfirst ? dim3(N / GG_BN, n_mtiles) : dim3(n_mtiles, N / GG_BN);
  static const int bk_env = []{
    const char* v = getenv("AMD_OPS_GG_BK");
    return v ? atoi(v) : 64;
  }();
  if (K % 128 == 0 && bk_env == 128) {
    const size_t smem = 2 * GG_BM * 128 * 2;
    hipLaunchKernelGGL((grouped_gemm_nt_kernel<128>), grid, dim3(256), smem,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<const bf16*>(w.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       tile_map.data_ptr<int>(), offs.data_ptr<int>(), ntp, K, N,
                       nfirst);
  } else {
    const size_t smem = 2 * GG_BM * 64 * 2;
    hipLaunchKernelGGL((grouped_gemm_nt_kernel<64>), grid, dim3(256), smem,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<const bf16*>(w.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       tile_map.data_ptr<int>(), offs.data_ptr<int>(), ntp, K, N,
                       nfirst);
  }
  HIP_CHECK_KERNEL();
  return y;
}

// ===========================================================================
// Grouped GEMM backward (the round-1 backward was a
// per-expert hipBLASLt loop — launch-bound at DeepSeek-style expert counts).
//   grouped_gemm_nn (dx): dx[m, k] = sum_n g[m, n] * w[e(m), n, k]
//     same tile structure as the forward, with the W tile staged TRANSPOSED
//     in LDS (wt[k][n]) so the B-fragment (8 consecutive n at fixed k) is a
//     contiguous ds_read_b128.
//   grouped_gemm_tn (dw): dw[e, n, k] = sum_{m in group e} g[m, n] * x[m, k]
//     grid (E, N/128, K/128); the m-loop stays inside the block so each dw
//     tile is written once — deterministic, no atomics.
// ===========================================================================

__global__ __launch_bounds__(256) void grouped_gemm_nn_kernel(
    const bf16* __restrict__ g, const bf16* __restrict__ w, bf16* __restrict__ dx,
    const int* __restrict__ tile_map, const int* __restrict__ offs,
    const int* __restrict__ n_tiles, int N, int K) {
  if (n_tiles != nullptr && (int)blockIdx.x >= n_tiles[0]) return;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* ga = smem;                       // [128 m][64 n]  16 KiB
  char* wt = smem + GG_BM * GG_BK * 2;   // [128 k][64 n]  16 KiB

  const int e = tile_map[2 * blockIdx.x];
  const int m0 = tile_map[2 * blockIdx.x + 1];
  const int m_end = offs[e + 1];
  const int k0 = blockIdx.y * GG_BN;     // output-k tile

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;
  const int kq = lane >> 4;

  const long wbase = (long)e * N * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  for (int n0 = 0; n0 < N; n0 += GG_BK) {
    {
      // g tile [128 m][64 n]: 128 rows x 8 vec-chunks = 1024 vectors
      const int row0 = tid / 8;
      const int c0 = (tid % 8) * 8;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = row0 + rr * 32;
        const int m = m0 + row;
        bf16x8 gv;
        if (m < m_end) {
          gv = *reinterpret_cast<const bf16x8*>(g + (long)m * N + n0 + c0);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) gv.v[j] = f2bf(0.f);
        }
        *reinterpret_cast<bf16x8*>(ga + gg_off(row, c0 * 2)) = gv;
      }
      // w tile [64 n rows][128 k]: read w[n][k-chunk], scatter into wt[k][n]
      const int nrow0 = tid / 16;
      const int kc0 = (tid % 16) * 8;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int nrow = nrow0 + rr * 16;
        bf16x8 wv = *reinterpret_cast<const bf16x8*>(
            w + wbase + (long)(n0 + nrow) * K + k0 + kc0);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          *reinterpret_cast<bf16*>(wt + gg_off(kc0 + j, nrow * 2)) = wv.v[j];
      }
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < GG_BK / 32; ++kk) {
      bf16x8v a[4], b[4];
      const int arow = (wid >> 1) * 64;
      const int brow = (wid & 1) * 64;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        a[i] = *reinterpret_cast<const bf16x8v*>(
            ga + gg_off(arow + i * 16 + l16, (kk * 32 + kq * 8) * 2));
        b[i] = *reinterpret_cast<const bf16x8v*>(
            wt + gg_off(brow + i * 16 + l16, (kk * 32 + kq * 8) * 2));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  const int mw = m0 + (wid >> 1) * 64;
  const int kw = k0 + (wid & 1) * 64;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mw + i * 16 + kq * 4 + r;
      if (m < m_end) {
#pragma unroll
        for (int j = 0; j < 4; ++j)
          dx[(long)m * K + kw + j * 16 + l16] = f2bf(acc[i][j][r]);
      }
    }
  }
}

__global__ __launch_bounds__(256) void grouped_gemm_tn_kernel(
    const bf16* __restrict__ gt, const bf16* __restrict__ xt, bf16* __restrict__ dw,
    const int* __restrict__ offs, int N, int K, long M) {
  // dw[e, n, k] = sum_{m in group e} gt[n, m] * xt[k, m]
  // Operands arrive PRE-TRANSPOSED ([N, M] / [K, M], m contiguous) so the
  // staging is plain vector loads/stores — the round-1 in-kernel scalar
  // transpose ran at 80 TF/s (profiles/moe_r2_kernel_stats.csv); this
  // NT-shaped plain-staged structure matches the forward kernel.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* ga = smem;                       // [128 n][64 m]
  char* xb = smem + GG_BM * GG_BK * 2;   // [128 k][64 m]

  // k-first grid (z = expert): consecutive blocks share the expert's g/x
  // row range, so its m-stream stays hot in the XCD L2s (same reasoning as
  // the NT n-first order; A/B via AMD_OPS_GG_TN_EFIRST=1 for the old order)
  const int e = blockIdx.z;
  const int n0 = blockIdx.y * GG_BN;
  const int k0 = blockIdx.x * GG_BN;
  const int m_start = offs[e], m_end = offs[e + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;
  const int kq = lane >> 4;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  const int row0 = tid / 8;          // 32 rows per pass (of 128)
  const int c0 = (tid % 8) * 8;      // 8 chunks cover 64 m
  const int mc0 = m_start & ~(GG_BK - 1);     // 64-aligned loop start
  for (int mc = mc0; mc < m_end; mc += GG_BK) {
    // plain staging (round-1 A/B: source-level double-buffering is a net
    // loss here — the extra register set costs occupancy; see NOTES_ROUND2)
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int row = row0 + rr * 32;
      const long m = (long)mc + c0;
      bf16x8 gv, xv;
      if (m >= m_start && m + 7 < m_end) {
        gv = *reinterpret_cast<const bf16x8*>(gt + (long)(n0 + row) * M + m);
        xv = *reinterpret_cast<const bf16x8*>(xt + (long)(k0 + row) * M + m);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const long mm = m + j;
          const bool ok = mm >= m_start && mm < m_end;
          gv.v[j] = ok ? gt[(long)(n0 + row) * M + mm] : f2bf(0.f);
          xv.v[j] = ok ? xt[(long)(k0 + row) * M + mm] : f2bf(0.f);
        }
      }
      *reinterpret_cast<bf16x8*>(ga + gg_off(row, c0 * 2)) = gv;
      *reinterpret_cast<bf16x8*>(xb + gg_off(row, c0 * 2)) = xv;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < GG_BK / 32; ++kk) {
      bf16x8v a[4], b[4];
      const int arow = (wid >> 1) * 64;    // n sub-tile
      const int brow = (wid & 1) * 64;     // k sub-tile
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        a[i] = *reinterpret_cast<const bf16x8v*>(
            ga + gg_off(arow + i * 16 + l16, (kk * 32 + kq * 8) * 2));
        b[i] = *reinterpret_cast<const bf16x8v*>(
            xb + gg_off(brow + i * 16 + l16, (kk * 32 + kq * 8) * 2));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  const long wbase = (long)e * N * K;
  const int nw = n0 + (wid >> 1) * 64;
  const int kw = k0 + (wid & 1) * 64;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int n = nw + i * 16 + kq * 4 + r;
#pragma unroll
      for (int j = 0; j < 4; ++j)
        dw[wbase + (long)n * K + kw + j * 16 + l16] = f2bf(acc[i][j][r]);
    }
  }
}

at::Tensor grouped_gemm_nn(const at::Tensor& g, const at::Tensor& w,
                           const at::Tensor& offs, const at::Tensor& tile_map,
                           const std::optional<at::Tensor>& n_tiles) {
  TORCH_CHECK(g.is_cuda() && g.dim() == 2 && g.scalar_type() == at::kBFloat16,
              "grouped_gemm_nn: g [M,N] bf16");
  TORCH_CHECK(w.dim() == 3 && w.scalar_type() == at::kBFloat16, "w [E,N,K] bf16");
  const long M = g.size(0);
  const int N = g.size(1), K = w.size(2);
  TORCH_CHECK(N == w.size(1), "N mismatch");
  TORCH_CHECK(N % GG_BK == 0 && K % GG_BN == 0, "need N%64==0, K%128==0");
  auto dx = at::empty({M, (long)K}, g.options());
  const int n_mtiles = tile_map.size(0);
  if (n_mtiles == 0 || M == 0) return dx;
  const dim3 grid(n_mtiles, K / GG_BN);
  const size_t smem = 2 * GG_BM * GG_BK * 2;
  auto stream = c10::hip::getCurrentHIPStream();
  const int* ntp = n_tiles.has_value() ? n_tiles->data_ptr<int>() : nullptr;
  hipLaunchKernelGGL(grouped_gemm_nn_kernel, grid, dim3(256), smem, stream.stream(),
                     reinterpret_cast<const bf16*>(g.data_ptr()),
                     reinterpret_cast<const bf16*>(w.data_ptr()),
                     reinterpret_cast<bf16*>(dx.data_ptr()),
                     tile_map.data_ptr<int>(), offs.data_ptr<int>(), ntp, N, K);
  HIP_CHECK_KERNEL();
  return dx;
}

at::Tensor grouped_gemm_tn(const at::Tensor& gt, const at::Tensor& xt,
                           const at::Tensor& offs, int64_t E) {
  TORCH_CHECK(gt.is_cuda() && gt.dim() == 2 && xt.dim() == 2 &&
                  gt.scalar_type() == at::kBFloat16,
              "grouped_gemm_tn: gt [N,M], xt [K,M] bf16 (pre-transposed)");
  const int N = gt.size(0), K = xt.size(0);
  const long M = gt.size(1);
  TORCH_CHECK(xt.size(1) == M, "M mismatch");
  TORCH_CHECK(N % GG_BN == 0 && K % GG_BN == 0, "need N%128==0, K%128==0");
  auto dw = at::empty({E, (long)N, (long)K}, gt.options());
  const dim3 grid(K / GG_BN, N / GG_BN, (unsigned)E);
  const size_t smem = 2 * GG_BM * GG_BK * 2;
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(grouped_gemm_tn_kernel, grid, dim3(256), smem, stream.stream(),
                     reinterpret_cast<const bf16*>(gt.data_ptr()),
                     reinterpret_cast<const bf16*>(xt.data_ptr()),
                     reinterpret_cast<bf16*>(dw.data_ptr()),
                     offs.data_ptr<int>(), N, K, M);
  HIP_CHECK_KERNEL();
  return dw;
}

// ---- bf16 tiled transpose (for the TN backward's pre-transposed operands;
// torch's strided transpose measured 6x off HBM roofline on [131072, 2048])
// 64x64 tiles through LDS: coalesced vector reads AND writes.
__global__ __launch_bounds__(256) void transpose_bf16_kernel(
    const bf16* __restrict__ in, bf16* __restrict__ out, int R, int C) {
  __shared__ bf16 tile[64][64 + 8];   // +8 bf16 pad kills bank conflicts
  const int tr0 = blockIdx.y * 64;    // input row tile
  const int tc0 = blockIdx.x * 64;    // input col tile
  const int tid = threadIdx.x;
  // read: 64 rows x 8 vectors of 8 -> 512 vector reads, 2 per thread
  for (int idx = tid; idx < 64 * 8; idx += 256) {
    const int r = idx / 8, c = (idx % 8) * 8;
    const int gr = tr0 + r, gc = tc0 + c;
    if (gr < R) {
      bf16x8 v;
      if (gc + 7 < C) {
        v = *reinterpret_cast<const bf16x8*>(in + (long)gr * C + gc);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v.v[j] = (gc + j < C) ? in[(long)gr * C + gc + j] : f2bf(0.f);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) tile[c + j][r] = v.v[j];
    }
  }
  __syncthreads();
  // write: transposed rows (= input cols), coalesced vectors
  for (int idx = tid; idx < 64 * 8; idx += 256) {
    const int r = idx / 8, c = (idx % 8) * 8;
    const int gr = tc0 + r, gc = tr0 + c;   // output [C, R]
    if (gr < C && gc < R) {
      bf16x8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = tile[r][c + j];
      if (gc + 7 < R) {
        *reinterpret_cast<bf16x8*>(out + (long)gr * R + gc) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (gc + j < R) out[(long)gr * R + gc + j] = v.v[j];
      }
    }
  }
}

at::Tensor transpose_bf16(const at::Tensor& in) {
  TORCH_CHECK(in.is_cuda() && in.dim() == 2 && in.scalar_type() == at::kBFloat16 &&
              in.is_contiguous(), "transpose_bf16: contiguous 2-D bf16");
  const int R = in.size(0), C = in.size(1);
  auto out = at::empty({C, R}, in.options());
  const dim3 grid((C + 63) / 64, (R + 63) / 64);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(in.data_ptr()),
                     reinterpret_cast<bf16*>(out.data_ptr()), R, C);
  HIP_CHECK_KERNEL();
  return out;
}

// ---- fused permute / unpermute -------------------------------------------
// gather: y[i, :] = x[src[i], :]
__global__ void permute_gather_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                                      const int* __restrict__ src, int H, long M) {
  const long i = blockIdx.x;
  const bf16* xr = x + (long)src[i] * H;
  bf16* yr = y + i * (long)H;
  for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8)
    *reinterpret_cast<bf16x8*>(yr + c) = *reinterpret_cast<const bf16x8*>(xr + c);
}

// combine: out[t, :] = sum_k probs[t, k] * yp[pos[t*K + k], :]
__global__ void unpermute_combine_kernel(const bf16* __restrict__ yp,
                                         bf16* __restrict__ out,
                                         const int* __restrict__ pos,
                                         const float* __restrict__ probs,
                                         int H, int topk, long T) {
  const long t = blockIdx.x;
  bf16* orow = out + t * (long)H;
  for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
    float accv[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int k = 0; k < topk; ++k) {
      const float p = probs[t * topk + k];
      const bf16x8 v = *reinterpret_cast<const bf16x8*>(
          yp + (long)pos[t * topk + k] * H + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) accv[j] += p * bf2f(v.v[j]);
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = f2bf(accv[j]);
    *reinterpret_cast<bf16x8*>(orow + c) = o;
  }
}

at::Tensor permute_gather(const at::Tensor& x, const at::Tensor& src) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.size(1) % 8 == 0,
              "permute_gather: bf16 [T,H], H%8==0");
  const long M = src.size(0);
  const int H = x.size(1);
  auto y = at::empty({M, (long)H}, x.options());
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(permute_gather_kernel, dim3(M), dim3(std::min(256, H / 8)), 0,
                     stream.stream(), reinterpret_cast<const bf16*>(x.data_ptr()),
                     reinterpret_cast<bf16*>(y.data_ptr()), src.data_ptr<int>(), H, M);
  HIP_CHECK_KERNEL();
  return y;
}

at::Tensor unpermute_combine(const at::Tensor& yp, const at::Tensor& pos,
                             const at::Tensor& probs) {
  const long T = probs.size(0);
  const int topk = probs.size(1);
  const int H = yp.size(1);
  auto out = at::empty({T, (long)H}, yp.options());
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(unpermute_combine_kernel, dim3(T), dim3(std::min(256, H / 8)), 0,
                     stream.stream(), reinterpret_cast<const bf16*>(yp.data_ptr()),
                     reinterpret_cast<bf16*>(out.data_ptr()), pos.data_ptr<int>(),
                     probs.data_ptr<float>(), H, topk, T);
  HIP_CHECK_KERNEL();
  return out;
}

// ===========================================================================
// fp8 grouped GEMM forward (SURVEY §2.5 last gap: the reference's MXFP8
// grouped path, moe/fp8_utils.py). Tensorwise-scaled OCP e4m3 operands,
// __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8 (bf16 MFMA rate, HALF the
// HBM/LDS bytes — the bf16 grouped kernels are memory-wait-bound per
// profiles/gg_r2_pmc_counters.csv, so fp8 wins on bandwidth, not rate).
// BK = 128 fp8 bytes per stage: rows are 128 B exactly like the bf16
// tiles, so the proven gg_off swizzle carries over unchanged.
// ===========================================================================

typedef long fp8x8;   // 8 packed e4m3 bytes = one MFMA operand

__global__ __launch_bounds__(256) void grouped_gemm_nt_fp8_kernel(
    const unsigned char* __restrict__ x, const unsigned char* __restrict__ w,
    bf16* __restrict__ y, const int* __restrict__ tile_map,
    const int* __restrict__ offs, const int* __restrict__ n_tiles,
    int K, int N, const float* __restrict__ scale_p, int nfirst) {
  const int tb = nfirst ? (int)blockIdx.y : (int)blockIdx.x;
  const int nb = nfirst ? (int)blockIdx.x : (int)blockIdx.y;
  if (n_tiles != nullptr && tb >= n_tiles[0]) return;
  const float scale = scale_p[0];
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* xa = smem;                  // [128 m][128 k-bytes] 16 KiB
  char* wb = smem + GG_BM * 128;

  const int e = tile_map[2 * tb];
  const int m0 = tile_map[2 * tb + 1];
  const int m_end = offs[e + 1];
  const int n0 = nb * GG_BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;
  const int kq = lane >> 4;

  const long wbase = (long)e * N * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  const int row0 = tid / 8;          // 8 x 16-B chunks per 128-B row
  const int c0 = (tid % 8) * 16;

  for (int k0 = 0; k0 < K; k0 += 128) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int row = row0 + rr * 32;
      const int m = m0 + row;
      int4 xv;
      if (m < m_end) {
        xv = *reinterpret_cast<const int4*>(x + (long)m * K + k0 + c0);
      } else {
        xv = make_int4(0, 0, 0, 0);
      }
      *reinterpret_cast<int4*>(xa + gg_off(row, c0)) = xv;
      int4 wv = *reinterpret_cast<const int4*>(
          w + wbase + (long)(n0 + row) * K + k0 + c0);
      *reinterpret_cast<int4*>(wb + gg_off(row, c0)) = wv;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      fp8x8 a[4], b[4];
      const int arow = (wid >> 1) * 64;
      const int brow = (wid & 1) * 64;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        a[i] = *reinterpret_cast<const fp8x8*>(
            xa + gg_off(arow + i * 16 + l16, kk * 32 + kq * 8));
        b[i] = *reinterpret_cast<const fp8x8*>(
            wb + gg_off(brow + i * 16 + l16, kk * 32 + kq * 8));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  const int mw = m0 + (wid >> 1) * 64;
  const int nw = n0 + (wid & 1) * 64;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mw + i * 16 + kq * 4 + r;
      if (m < m_end) {
#pragma unroll
        for (int j = 0; j < 4; ++j)
          y[(long)m * N + nw + j * 16 + l16] = f2bf(acc[i][j][r] * scale);
      }
    }
  }
}

// 256x256 fp8 big-tile (8 waves, 64 KiB LDS): the production MoE tile.
__global__ __launch_bounds__(512, 1) void grouped_gemm_nt_fp8_big_kernel(
    const unsigned char* __restrict__ x, const unsigned char* __restrict__ w,
    bf16* __restrict__ y, const int* __restrict__ tile_map,
    const int* __restrict__ offs, const int* __restrict__ n_tiles,
    int K, int N, const float* __restrict__ scale_p, int nfirst) {
  const int tb = nfirst ? (int)blockIdx.y : (int)blockIdx.x;
  const int nb = nfirst ? (int)blockIdx.x : (int)blockIdx.y;
  if (n_tiles != nullptr && tb >= n_tiles[0]) return;
  const float scale = scale_p[0];
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* xa = smem;                    // [256 m][128 k-bytes] 32 KiB
  char* wb = smem + 256 * 128;

  const int e = tile_map[2 * tb];
  const int m0 = tile_map[2 * tb + 1];
  const int m_end = offs[e + 1];
  const int n0 = nb * 256;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l16 = lane & 15;
  const int kq = lane >> 4;

  const long wbase = (long)e * N * K;

  f32x4 acc[4][8];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  const int row0 = tid / 8;            // 64 rows per pass (of 256)
  const int c0 = (tid % 8) * 16;

  for (int k0 = 0; k0 < K; k0 += 128) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int row = row0 + rr * 64;
      const int m = m0 + row;
      int4 xv;
      if (m < m_end) {
        xv = *reinterpret_cast<const int4*>(x + (long)m * K + k0 + c0);
      } else {
        xv = make_int4(0, 0, 0, 0);
      }
      *reinterpret_cast<int4*>(xa + gg_off(row, c0)) = xv;
      int4 wv = *reinterpret_cast<const int4*>(
          w + wbase + (long)(n0 + row) * K + k0 + c0);
      *reinterpret_cast<int4*>(wb + gg_off(row, c0)) = wv;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      fp8x8 a[4], b[8];
      const int arow = (wid >> 1) * 64;
      const int brow = (wid & 1) * 128;
#pragma unroll
      for (int i = 0; i < 4; ++i)
        a[i] = *reinterpret_cast<const fp8x8*>(
            xa + gg_off(arow + i * 16 + l16, kk * 32 + kq * 8));
#pragma unroll
      for (int j = 0; j < 8; ++j)
        b[j] = *reinterpret_cast<const fp8x8*>(
            wb + gg_off(brow + j * 16 + l16, kk * 32 + kq * 8));
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  const int mw = m0 + (wid >> 1) * 64;
  const int nw = n0 + (wid & 1) * 128;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mw + i * 16 + kq * 4 + r;
      if (m < m_end) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          y[(long)m * N + nw + j * 16 + l16] = f2bf(acc[i][j][r] * scale);
      }
    }
  }
}

at::Tensor grouped_gemm_nt_fp8(const at::Tensor& x8, const at::Tensor& w8,
                               const at::Tensor& offs, const at::Tensor& tile_map,
                               const at::Tensor& scale,
                               const std::optional<at::Tensor>& n_tiles,
                               int64_t bm) {
  TORCH_CHECK(scale.is_cuda() && scale.scalar_type() == at::kFloat && scale.numel() == 1,
              "scale: 1-elem float32 device tensor (combined sx*sw dequant)");
  TORCH_CHECK(x8.is_cuda() && x8.dim() == 2 && x8.element_size() == 1,
              "grouped_gemm_nt_fp8: x8 [M,K] e4m3 bytes");
  TORCH_CHECK(w8.dim() == 3 && w8.element_size() == 1, "w8 [E,N,K] e4m3 bytes");
  const long M = x8.size(0);
  const int K = x8.size(1), N = w8.size(1);
  TORCH_CHECK(K == w8.size(2), "K mismatch");
  TORCH_CHECK(K % 128 == 0 && N % GG_BN == 0, "need K%128==0, N%128==0");
  auto y = at::empty({M, (long)N},
                     x8.options().dtype(at::kBFloat16));
  const int n_mtiles = tile_map.size(0);
  if (n_mtiles == 0 || M == 0) return y;
  auto stream = c10::hip::getCurrentHIPStream();
  const int* ntp = n_tiles.has_value() ? n_tiles->data_ptr<int>() : nullptr;
  if (bm == 256) {
    TORCH_CHECK(N % 256 == 0, "big-tile fp8 nt needs N%256==0");
    const int nfirstb = gg_nfirst();
    const dim3 gridb = nfirstb ? dim3(N / 256, n_mtiles) : dim3(n_mtiles, N / 256);
    hipLaunchKernelGGL(grouped_gemm_nt_fp8_big_kernel, gridb, dim3(512),
                       2 * 256 * 128, stream.stream(),
                       reinterpret_cast<const unsigned char*>(x8.data_ptr()),
                       reinterpret_cast<const unsigned char*>(w8.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()),
                       tile_map.data_ptr<int>(), offs.data_ptr<int>(), ntp, K, N,
                       scale.data_ptr<float>(), nfirstb);
    HIP_CHECK_KERNEL();
    return y;
  }
  const int nfirst = gg_nfirst();
  const dim3 grid = nfirst ? dim3(N / GG_BN, n_mtiles) : dim3(n_mtiles, N / GG_BN);
  hipLaunchKernelGGL(grouped_gemm_nt_fp8_kernel, grid, dim3(256),
                     2 * GG_BM * 128, stream.stream(),
                     reinterpret_cast<const unsigned char*>(x8.data_ptr()),
                       reinterpret_cast<const unsigned char*>(w8.data_ptr()),
                     reinterpret_cast<bf16*>(y.data_ptr()),
                     tile_map.data_ptr<int>(), offs.data_ptr<int>(), ntp, K, N,
                     scale.data_ptr<float>(), nfirst);
  HIP_CHECK_KERNEL();
  return y;
}

}  // namespace amd_ops
