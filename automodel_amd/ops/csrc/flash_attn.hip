// Flash attention forward (CDNA4, gfx950) — BSHD, causal, GQA, D=128.
//
// MI355X-native replacement for the reference's TE DotProductAttention /
// flash-attn backends. Structure follows the CDNA4
// guide's fused-attention recipe (cdna_hip_programming.md Appendix B):
//   * swapped QK^T — mfma_f32_32x32x16_bf16 computing mfma(K, Q) so each
//     lane holds a P column for ONE q row -> softmax is in-register
//     (16 regs + one __shfl_xor(32) cross-half exchange)
//   * online softmax (running m, l per q row)
//   * P -> bf16 via packed cvt + __builtin_amdgcn_permlane32_swap to build
//     the PV A-fragment without LDS round trips (guide T12)
//   * K tile LDS-staged with XOR swizzle (guide T2 / Guideline 4: row-major
//     [32][128] bf16 is a 16-way ds_read_b128 conflict without it)
//   * V tile staged TRANSPOSED in LDS (vt[d][k]) with its own XOR swizzle
//     so PV B-fragments are contiguous ds_read_b128
//
// Workgroup: 4 waves, each owning QBLK=32 q rows (block tile = 128 rows),
// sharing the K/V LDS tiles; KVBLK=32 per iteration.
//
// Backward is currently a composite of hipBLASLt GEMMs driven from python
// (ops/attention.py); the fused HIP backward kernel is the next milestone.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

typedef __bf16 bf16x8_v __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

#define QBLK 32
#define KVBLK 32
#define FA_WAVES 4
#define FA_BLOCK (FA_WAVES * WAVE_SIZE)

// K tile: [KVBLK][D] bf16 row-major, 256 B rows, XOR-swizzled by ((row&7)<<4).
__device__ __forceinline__ int k_lds_off(int row, int byte_in_row) {
  return row * 256 + (byte_in_row ^ ((row & 7) << 4));
}
// V^T tile: [D][KVBLK] bf16, 64 B rows, XOR-swizzled by ((row&3)<<4).
__device__ __forceinline__ int vt_lds_off(int d, int byte_in_row) {
  return d * 64 + (byte_in_row ^ ((d & 3) << 4));
}

template <int D>
__global__ __launch_bounds__(FA_BLOCK) void flash_fwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k, const bf16* __restrict__ v,
    bf16* __restrict__ out, float* __restrict__ lse, int B, int S, int Hq, int Hk,
    float scale, bool causal) {
  static_assert(D == 128, "flash_fwd: D=128 only for now");
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                       // KVBLK*D*2 = 8 KiB
  char* vt_lds = smem + KVBLK * D * 2;      // D*KVBLK*2 = 8 KiB
  float* bcast = reinterpret_cast<float*>(smem + 2 * KVBLK * D * 2);  // FA_WAVES*32

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE_SIZE - 1);
  const int wid = tid / WAVE_SIZE;
  const int col = lane & 31;           // q row within wave tile (QK layout)
  const int half = lane >> 5;

  const int qblk0 = blockIdx.x * (FA_WAVES * QBLK);
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (Hq / Hk);

  const long q_base = (((long)b * S) * Hq + h) * D;        // + s*Hq*D
  const long kv_base = (((long)b * S) * Hk + kvh) * D;     // + s*Hk*D
  const long q_row_stride = (long)Hq * D;
  const long kv_row_stride = (long)Hk * D;

  const int q0 = qblk0 + wid * QBLK;   // this wave's first q row

  // ---- load Q fragments to registers: lane holds Q[q0+col][8*half + j + 16*c]
  bf16x8_v qfrag[D / 16];
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    const bf16* src = q + q_base + (long)(q0 + col) * q_row_stride + c * 16 + half * 8;
    qfrag[c] = *reinterpret_cast<const bf16x8_v*>(src);
  }

  // ---- accumulators
  f32x16 o_acc[D / 32];
#pragma unroll
  for (int t = 0; t < D / 32; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[t][r] = 0.f;
  float m_run = -1e30f, l_run = 0.f;

  const int q_block_max = qblk0 + FA_WAVES * QBLK - 1;
  const int n_tiles = causal ? (min(q_block_max, S - 1) / KVBLK + 1) : (S + KVBLK - 1) / KVBLK;

  for (int jt = 0; jt < n_tiles; ++jt) {
    const int k0 = jt * KVBLK;
    // ---- stage K tile (swizzled) and V^T tile cooperatively
    {
      // 256 threads, tile = 32 rows x 128 cols: thread t -> row t/16+{0,16}, col (t%16)*8
      const int r0 = tid / 16, c0 = (tid % 16) * 8;
#pragma unroll
      for (int rr = 0; rr < 2; ++rr) {
        const int row = r0 + rr * 16;
        const int ks = k0 + row;
        bf16x8 kv8 = *reinterpret_cast<const bf16x8*>(k + kv_base + (long)ks * kv_row_stride + c0);
        *reinterpret_cast<bf16x8*>(k_lds + k_lds_off(row, c0 * 2)) = kv8;
        bf16x8 vv8 = *reinterpret_cast<const bf16x8*>(v + kv_base + (long)ks * kv_row_stride + c0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {  // transpose store: vt[d][k]
          *reinterpret_cast<bf16*>(vt_lds + vt_lds_off(c0 + j, row * 2)) = vv8.v[j];
        }
      }
    }
    __syncthreads();

    const bool tile_live = !causal || (k0 <= q0 + QBLK - 1);
    float alpha = 1.f;
    bf16x8_v pa0, pa1;
    if (tile_live) {
      // ---- QK^T swapped: P[k][q] = sum_d K[k][d] * Q[q][d]
      f32x16 p;
#pragma unroll
      for (int r = 0; r < 16; ++r) p[r] = 0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        // A fragment: K[l&31][8*half + j] at d-chunk c
        bf16x8_v ka = *reinterpret_cast<const bf16x8_v*>(
            k_lds + k_lds_off(col, (c * 16 + half * 8) * 2));
        p = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[c], p, 0, 0, 0);
      }

      // ---- masked online softmax (lane owns q row q0+col, ks (r&3)+8*(r>>2)+4*half)
      const int qg = q0 + col;
      float s_val[16];
      float tile_max = -1e30f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kg = k0 + (r & 3) + 8 * (r >> 2) + 4 * half;
        float sv = p[r] * scale;
        if ((causal && kg > qg) || kg >= S) sv = -1e30f;
        s_val[r] = sv;
        tile_max = fmaxf(tile_max, sv);
      }
      tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32));
      const float m_new = fmaxf(m_run, tile_max);
      alpha = __expf(m_run - m_new);
      float psum = 0.f;
      float pv[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        pv[r] = __expf(s_val[r] - m_new);
        psum += pv[r];
      }
      psum += __shfl_xor(psum, 32);
      l_run = l_run * alpha + psum;
      m_run = m_new;

      // ---- P f32 -> bf16 packed pairs, permlane32_swap into A-fragment layout
      unsigned int pk[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        unsigned lo = __builtin_bit_cast(unsigned short, f2bf(pv[2 * i]));
        unsigned hi = __builtin_bit_cast(unsigned short, f2bf(pv[2 * i + 1]));
        pk[i] = lo | (hi << 16);
      }
      // fragment 0: k 0..15  <- regs 0..7 ; fragment 1: k 16..31 <- regs 8..15
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        auto r02 = __builtin_amdgcn_permlane32_swap(pk[4 * i + 0], pk[4 * i + 2], false, false);
        auto r13 = __builtin_amdgcn_permlane32_swap(pk[4 * i + 1], pk[4 * i + 3], false, false);
        unsigned frag[4] = {(unsigned)r02[0], (unsigned)r13[0], (unsigned)r02[1], (unsigned)r13[1]};
        if (i == 0) pa0 = __builtin_bit_cast(bf16x8_v, *reinterpret_cast<ulonglong2*>(frag));
        else pa1 = __builtin_bit_cast(bf16x8_v, *reinterpret_cast<ulonglong2*>(frag));
      }

      // ---- broadcast alpha to O layout via LDS
      if (lane < 32) bcast[wid * 32 + lane] = alpha;
    }
    // (waves with dead tiles skip compute but still hit the barriers)
    __syncthreads();

    if (tile_live) {
      // ---- rescale O by alpha (per q row: row = (r&3) + 8*(r>>2) + 4*half)
#pragma unroll
      for (int t = 0; t < D / 32; ++t) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int row = (r & 3) + 8 * (r >> 2) + 4 * half;
          o_acc[t][r] *= bcast[wid * 32 + row];
        }
      }
#pragma unroll
      for (int t = 0; t < D / 32; ++t) {
        // B fragments: vt[d = t*32 + (l&31)][k], k chunks of 8
        bf16x8_v vb0 = *reinterpret_cast<const bf16x8_v*>(
            vt_lds + vt_lds_off(t * 32 + col, (half * 8) * 2));
        bf16x8_v vb1 = *reinterpret_cast<const bf16x8_v*>(
            vt_lds + vt_lds_off(t * 32 + col, (16 + half * 8) * 2));
        o_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa0, vb0, o_acc[t], 0, 0, 0);
        o_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa1, vb1, o_acc[t], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: normalize by l, write O and LSE
  if (lane < 32) bcast[wid * 32 + lane] = (l_run > 0.f) ? 1.f / l_run : 0.f;
  __syncthreads();

  const int qg = q0 + col;
  if (lane < 32 && qg < S) {
    lse[((long)b * Hq + h) * S + qg] = (l_run > 0.f) ? m_run + __logf(l_run) : -1e30f;
  }
#pragma unroll
  for (int t = 0; t < D / 32; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * half;  // q row in wave tile
      const int qrow = q0 + row;
      if (qrow < S) {
        const float inv_l = bcast[wid * 32 + row];
        out[q_base + (long)qrow * q_row_stride + t * 32 + col] = f2bf(o_acc[t][r] * inv_l);
      }
    }
  }
}

std::tuple<at::Tensor, at::Tensor> flash_attn_fwd(const at::Tensor& q, const at::Tensor& k,
                                                  const at::Tensor& v, double scale,
                                                  bool causal) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.scalar_type() == at::kBFloat16,
              "flash_attn_fwd: q must be [B,S,Hq,D] bf16");
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hk = k.size(2);
  TORCH_CHECK(D == 128, "flash_attn_fwd: only D=128 supported, got ", D);
  TORCH_CHECK(Hq % Hk == 0, "flash_attn_fwd: Hq must be divisible by Hk");
  TORCH_CHECK(S % (FA_WAVES * QBLK) == 0, "flash_attn_fwd: S must be a multiple of 128");
  auto out = at::empty_like(q);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  const dim3 grid(S / (FA_WAVES * QBLK), Hq, B);
  const size_t smem = 2 * KVBLK * 128 * 2 + FA_WAVES * 32 * sizeof(float);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((flash_fwd_kernel<128>), grid, dim3(FA_BLOCK), smem, stream.stream(),
                     reinterpret_cast<const bf16*>(q.data_ptr()),
                     reinterpret_cast<const bf16*>(k.data_ptr()),
                     reinterpret_cast<const bf16*>(v.data_ptr()),
                     reinterpret_cast<bf16*>(out.data_ptr()), lse.data_ptr<float>(),
                     B, S, Hq, Hk, (float)scale, causal);
  HIP_CHECK_KERNEL();
  return {out, lse};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> flash_attn_bwd(
    const at::Tensor& dout, const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& o, const at::Tensor& lse, double scale, bool causal) {
  TORCH_CHECK(false,
              "flash_attn_bwd HIP kernel not built yet — python wrapper uses the "
              "GEMM-composite backward (ops/attention.py)");
}

// ---- MFMA layout self-test: d[32,32] = a[32,16] @ b[16,32] via one mfma.
__global__ void mfma_probe_kernel(const bf16* a, const bf16* b, float* d) {
  const int lane = threadIdx.x & 63;
  const int half = lane >> 5;
  bf16x8_v af, bf_;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = (__bf16)a[(lane & 31) * 16 + half * 8 + j];   // A[i=l&31][k=8*half+j]
    bf_[j] = (__bf16)b[(half * 8 + j) * 32 + (lane & 31)];  // B[k=8*half+j][j=l&31]
  }
  f32x16 acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf_, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * half;
    d[row * 32 + (lane & 31)] = acc[r];
  }
}

at::Tensor mfma_probe(const at::Tensor& a, const at::Tensor& b) {
  TORCH_CHECK(a.is_cuda() && a.sizes() == at::IntArrayRef({32, 16}), "a must be [32,16]");
  TORCH_CHECK(b.sizes() == at::IntArrayRef({16, 32}), "b must be [16,32]");
  auto d = at::empty({32, 32}, a.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(a.data_ptr()),
                     reinterpret_cast<const bf16*>(b.data_ptr()), d.data_ptr<float>());
  HIP_CHECK_KERNEL();
  return d;
}

}  // namespace amd_ops
