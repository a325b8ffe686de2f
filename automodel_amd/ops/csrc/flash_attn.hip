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

// K tile: [KVBLK][D] bf16 row-major, 256 B rows. A 256-B row starts every
// 64 banks, so ALL rows share bank 0 without a swizzle; ((row&15)<<4)
// spreads a 16-lane ds_read_b128 group over all 16 slots -> conflict-free
// (guide Guideline 4). PMC before the fix: SQ_LDS_BANK_CONFLICT 2.6e11 in
// the dkv kernel alone (profiles/r1_pmc_counters.csv).
__device__ __forceinline__ int k_lds_off(int row, int byte_in_row) {
  return row * 256 + (byte_in_row ^ ((row & 15) << 4));
}
// V^T tile: [D][KVBLK] bf16, 64 B rows (16 words). Rows d and d+4 share the
// same 16-word bank window (16*d mod 64); rotating the 16-B slot by
// (d>>2)&3 separates the four colliding rows. (The previous (d&3) rotation
// only moved rows that never collided.)
__device__ __forceinline__ int vt_lds_off(int d, int byte_in_row) {
  // ((d>>2)^(d>>3))&3: the read-colliding set {d, d+4, d+8, d+12} still maps
  // to 4 distinct slots (conflict-free b128 reads) AND the transpose-store
  // writers (d stepping 8) now spread over all 4 slots (8-way -> 4-way).
  return d * 64 + (byte_in_row ^ ((((d >> 2) ^ (d >> 3)) & 3) << 4));
}

template <int D, int WAVES>
__global__ __launch_bounds__(WAVES * WAVE_SIZE, 2) void flash_fwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k, const bf16* __restrict__ v,
    bf16* __restrict__ out, float* __restrict__ lse, int B, int Sq, int Skv, int Hq,
    int Hk, int q_start, float scale, bool causal) {
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

  const int qblk0 = blockIdx.x * (WAVES * QBLK);
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (Hq / Hk);

  const long q_base = (((long)b * Sq) * Hq + h) * D;       // + s*Hq*D
  const long kv_base = (((long)b * Skv) * Hk + kvh) * D;   // + s*Hk*D
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

  const int q_block_max = q_start + qblk0 + WAVES * QBLK - 1;
  const int n_tiles = causal ? (min(q_block_max, Skv - 1) / KVBLK + 1)
                             : (Skv + KVBLK - 1) / KVBLK;

  for (int jt = 0; jt < n_tiles; ++jt) {
    const int k0 = jt * KVBLK;
    // ---- stage K tile (swizzled) and V^T tile cooperatively
    {
      // tile = 32 rows x 128 cols; 16 threads per row, WAVES*4 rows per pass
      const int c0 = (tid % 16) * 8;
#pragma unroll
      for (int row = tid / 16; row < KVBLK; row += WAVES * 4) {
        const int ks = min(k0 + row, Skv - 1);  // overhang rows masked later
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

    const bool tile_live = !causal || (k0 <= q_start + q0 + QBLK - 1);
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

      // ---- masked online softmax, IN PLACE in the p accumulator (register
      // budget: s_val/pv scratch arrays cost 32 VGPRs -> occupancy cliff)
      const int qg = q_start + q0 + col;
      float tile_max = -1e30f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kg = k0 + (r & 3) + 8 * (r >> 2) + 4 * half;
        float sv = p[r] * scale;
        if ((causal && kg > qg) || kg >= Skv) sv = -1e30f;
        p[r] = sv;
        tile_max = fmaxf(tile_max, sv);
      }
      tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32));
      const float m_new = fmaxf(m_run, tile_max);
      alpha = __expf(m_run - m_new);
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float e = __expf(p[r] - m_new);
        p[r] = e;
        psum += e;
      }
      psum += __shfl_xor(psum, 32);
      l_run = l_run * alpha + psum;
      m_run = m_new;

      // ---- P f32 -> bf16 packed pairs, permlane32_swap into A-fragment layout
      unsigned int pk[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        unsigned lo = __builtin_bit_cast(unsigned short, f2bf(p[2 * i]));
        unsigned hi = __builtin_bit_cast(unsigned short, f2bf(p[2 * i + 1]));
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
  if (lane < 32 && qg < Sq) {
    lse[((long)b * Hq + h) * Sq + qg] = (l_run > 0.f) ? m_run + __logf(l_run) : -1e30f;
  }
#pragma unroll
  for (int t = 0; t < D / 32; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * half;  // q row in wave tile
      const int qrow = q0 + row;
      if (qrow < Sq) {
        const float inv_l = bcast[wid * 32 + row];
        out[q_base + (long)qrow * q_row_stride + t * 32 + col] = f2bf(o_acc[t][r] * inv_l);
      }
    }
  }
}

std::tuple<at::Tensor, at::Tensor> flash_attn_fwd(const at::Tensor& q, const at::Tensor& k,
                                                  const at::Tensor& v, double scale,
                                                  bool causal, int64_t q_start) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.scalar_type() == at::kBFloat16,
              "flash_attn_fwd: q must be [B,S,Hq,D] bf16");
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1), Hk = k.size(2);
  TORCH_CHECK(D == 128, "flash_attn_fwd: only D=128 supported, got ", D);
  TORCH_CHECK(Hq % Hk == 0, "flash_attn_fwd: Hq must be divisible by Hk");
  TORCH_CHECK(Sq % (FA_WAVES * QBLK) == 0, "flash_attn_fwd: Sq must be a multiple of 128");
  TORCH_CHECK(Skv % KVBLK == 0, "flash_attn_fwd: Skv must be a multiple of 32");
  auto out = at::empty_like(q);
  auto lse = at::empty({B, Hq, Sq}, q.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  // 8-wave blocks amortize the shared K/V staging over twice the waves;
  // 4-wave fallback keeps small/CP-chunk shapes (Sq % 256 != 0) working.
  if (Sq % (8 * QBLK) == 0) {
    const dim3 grid(Sq / (8 * QBLK), Hq, B);
    const size_t smem = 2 * KVBLK * 128 * 2 + 8 * 32 * sizeof(float);
    hipLaunchKernelGGL((flash_fwd_kernel<128, 8>), grid, dim3(512), smem, stream.stream(),
                       reinterpret_cast<const bf16*>(q.data_ptr()),
                       reinterpret_cast<const bf16*>(k.data_ptr()),
                       reinterpret_cast<const bf16*>(v.data_ptr()),
                       reinterpret_cast<bf16*>(out.data_ptr()), lse.data_ptr<float>(),
                       B, Sq, Skv, Hq, Hk, (int)q_start, (float)scale, causal);
  } else {
    const dim3 grid(Sq / (4 * QBLK), Hq, B);
    const size_t smem = 2 * KVBLK * 128 * 2 + 4 * 32 * sizeof(float);
    hipLaunchKernelGGL((flash_fwd_kernel<128, 4>), grid, dim3(256), smem, stream.stream(),
                       reinterpret_cast<const bf16*>(q.data_ptr()),
                       reinterpret_cast<const bf16*>(k.data_ptr()),
                       reinterpret_cast<const bf16*>(v.data_ptr()),
                       reinterpret_cast<bf16*>(out.data_ptr()), lse.data_ptr<float>(),
                       B, Sq, Skv, Hq, Hk, (int)q_start, (float)scale, causal);
  }
  HIP_CHECK_KERNEL();
  return {out, lse};
}

// ===========================================================================
// Backward: flash-attention-2 style split.
//   delta kernel: delta[b,h,s] = rowsum(dO * O)
//   kernel A (kv-parallel): recompute S^T (swapped, lse/delta lane-local),
//     accumulate dK, dV; P^T/dS^T transposed through a per-wave LDS buffer;
//     f32 atomicAdd into dk/dv (GQA heads collapse onto the kv head).
//   kernel B (q-parallel): recompute S in [q][k] layout, accumulate dQ
//     (plain stores — q tiles are exclusive per block).
// Replaces the GEMM-composite python backward (profiles/bench8b round-1
// showed its masked_fill/exp/mul/f32-add chain at ~25% of step time).
// ===========================================================================

__global__ void fa_delta_kernel(const bf16* __restrict__ dout, const bf16* __restrict__ o,
                                float* __restrict__ delta, int S, int Hq, long rows) {
  // row r = ((b*S + s)*Hq + h); one wave per row, D=128 -> 2 elems/lane
  const long row = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE_SIZE;
  if (row >= rows) return;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const bf16* dp = dout + row * 128 + lane * 2;
  const bf16* op = o + row * 128 + lane * 2;
  float acc = bf2f(dp[0]) * bf2f(op[0]) + bf2f(dp[1]) * bf2f(op[1]);
  acc = wave_reduce_sum(acc);
  if (lane == 0) {
    // delta layout [B,Hq,S]: row -> (b, s, h)
    const long h = row % Hq;
    const long bs = row / Hq;
    const long b = bs / S, s = bs % S;
    delta[(b * Hq + h) * S + s] = acc;
  }
}

// q/do row images: [32][256 B], XOR swizzle ((row&7)<<4) (same as k_lds_off)
// qt/dot images:   [128][64 B], XOR swizzle ((d&3)<<4)   (same as vt_lds_off)
// per-wave transpose buffer: [32][64 B] with ((row&3)<<4)
__device__ __forceinline__ int tb_off(int row, int byte_in_row) {
  return row * 64 + (byte_in_row ^ ((((row >> 2) ^ (row >> 3)) & 3) << 4));
}

#define FAB_WAVES 4

template <int D, int WAVES>
__global__ __launch_bounds__(WAVES * WAVE_SIZE, 2) void flash_bwd_dkv_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const float* __restrict__ lse, const float* __restrict__ delta,
    float* __restrict__ dk, float* __restrict__ dv, int B, int Sq, int Skv, int Hq,
    int Hk, int q_start, float scale, bool causal) {
  static_assert(D == 128);
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* q_rows = smem;                 // 32*256 = 8 KiB
  char* do_rows = smem + 8 * 1024;     // 8 KiB
  char* qt = smem + 16 * 1024;         // 128*64 = 8 KiB
  char* dot = smem + 24 * 1024;        // 8 KiB
  char* tbuf = smem + 32 * 1024;       // per-wave 2 KiB x4

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid / 64;
  const int col = lane & 31;
  const int half = lane >> 5;

  const int kvb = blockIdx.x * (WAVES * 32);
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (Hq / Hk);
  const int kv0 = kvb + wid * 32;      // this wave's kv rows

  const long q_base = (((long)b * Sq) * Hq + h) * D;
  const long kv_base = (((long)b * Skv) * Hk + kvh) * D;
  const long q_rs = (long)Hq * D, kv_rs = (long)Hk * D;
  const float* lse_row = lse + ((long)b * Hq + h) * Sq;
  const float* dlt_row = delta + ((long)b * Hq + h) * Sq;

  // K/V rows of this wave -> A fragments in registers
  bf16x8_v kfrag[D / 16], vfrag[D / 16];
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    const long off = kv_base + (long)(kv0 + col) * kv_rs + c * 16 + half * 8;
    kfrag[c] = *reinterpret_cast<const bf16x8_v*>(k + off);
    vfrag[c] = *reinterpret_cast<const bf16x8_v*>(v + off);
  }

  f32x16 dk_acc[D / 32], dv_acc[D / 32];
#pragma unroll
  for (int t = 0; t < D / 32; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) { dk_acc[t][r] = 0.f; dv_acc[t][r] = 0.f; }

  char* tb = tbuf + wid * 2048;
  const int jq_start = causal ? (kvb > q_start ? (kvb - q_start) / 32 : 0) : 0;

  for (int jq = jq_start; jq < Sq / 32; ++jq) {
    const int q0 = jq * 32;
    // ---- cooperative stage: q/do rows + transposed images
    {
      const int c0 = (tid % 16) * 8;   // 16 threads per row of 128
      for (int rr = tid / 16; rr < 32; rr += WAVES * 4) {
        bf16x8 qv = *reinterpret_cast<const bf16x8*>(q + q_base + (long)(q0 + rr) * q_rs + c0);
        bf16x8 dv8 = *reinterpret_cast<const bf16x8*>(dout + q_base + (long)(q0 + rr) * q_rs + c0);
        *reinterpret_cast<bf16x8*>(q_rows + k_lds_off(rr, c0 * 2)) = qv;
        *reinterpret_cast<bf16x8*>(do_rows + k_lds_off(rr, c0 * 2)) = dv8;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          *reinterpret_cast<bf16*>(qt + vt_lds_off(c0 + j, rr * 2)) = qv.v[j];
          *reinterpret_cast<bf16*>(dot + vt_lds_off(c0 + j, rr * 2)) = dv8.v[j];
        }
      }
    }
    __syncthreads();

    const bool live = !causal || (q_start + q0 + 31 >= kv0);
    if (live) {
      const int ql = q0 + col;          // local q row (lse/delta index)
      const int qg = q_start + ql;      // global position (mask)
      const float lse_q = lse_row[ql];
      const float dlt_q = dlt_row[ql];

      // ---- S^T = K Q^T (D rows = kv, cols = q)
      f32x16 st;
#pragma unroll
      for (int r = 0; r < 16; ++r) st[r] = 0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        bf16x8_v qb = *reinterpret_cast<const bf16x8_v*>(
            q_rows + k_lds_off(col, (c * 16 + half * 8) * 2));
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag[c], qb, st, 0, 0, 0);
      }
      // ---- dP^T = V dO^T
      f32x16 dpt;
#pragma unroll
      for (int r = 0; r < 16; ++r) dpt[r] = 0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        bf16x8_v db = *reinterpret_cast<const bf16x8_v*>(
            do_rows + k_lds_off(col, (c * 16 + half * 8) * 2));
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag[c], db, dpt, 0, 0, 0);
      }

      // in place: st becomes P^T, dpt becomes dS^T (register budget)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kg = kv0 + (r & 3) + 8 * (r >> 2) + 4 * half;
        const bool masked = (causal && kg > qg) || kg >= Skv;
        const float pv = masked ? 0.f : __expf(st[r] * scale - lse_q);
        st[r] = pv;
        dpt[r] = pv * (dpt[r] - dlt_q) * scale;
      }

      // ---- transpose P^T -> A-frags via wave-local LDS buffer, accumulate dV
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int krow = (r & 3) + 8 * (r >> 2) + 4 * half;
        *reinterpret_cast<bf16*>(tb + tb_off(krow, col * 2)) = f2bf(st[r]);
      }
      __builtin_amdgcn_wave_barrier();
#pragma unroll
      for (int c2 = 0; c2 < 2; ++c2) {
        bf16x8_v pa = *reinterpret_cast<const bf16x8_v*>(
            tb + tb_off(col, (c2 * 16 + half * 8) * 2));
#pragma unroll
        for (int t = 0; t < D / 32; ++t) {
          bf16x8_v dob = *reinterpret_cast<const bf16x8_v*>(
              dot + vt_lds_off(t * 32 + col, (c2 * 16 + half * 8) * 2));
          dv_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, dob, dv_acc[t], 0, 0, 0);
        }
      }
      // ---- transpose dS^T, accumulate dK
      __builtin_amdgcn_wave_barrier();
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int krow = (r & 3) + 8 * (r >> 2) + 4 * half;
        *reinterpret_cast<bf16*>(tb + tb_off(krow, col * 2)) = f2bf(dpt[r]);
      }
      __builtin_amdgcn_wave_barrier();
#pragma unroll
      for (int c2 = 0; c2 < 2; ++c2) {
        bf16x8_v da = *reinterpret_cast<const bf16x8_v*>(
            tb + tb_off(col, (c2 * 16 + half * 8) * 2));
#pragma unroll
        for (int t = 0; t < D / 32; ++t) {
          bf16x8_v qb2 = *reinterpret_cast<const bf16x8_v*>(
              qt + vt_lds_off(t * 32 + col, (c2 * 16 + half * 8) * 2));
          dk_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, qb2, dk_acc[t], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: atomic accumulate into f32 dk/dv [B,S,Hk,D]
#pragma unroll
  for (int t = 0; t < D / 32; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = (r & 3) + 8 * (r >> 2) + 4 * half;
      const int kg = kv0 + krow;
      if (kg < Skv) {
        const long off = kv_base + (long)kg * kv_rs + t * 32 + col;
        atomicAdd(dk + off, dk_acc[t][r]);
        atomicAdd(dv + off, dv_acc[t][r]);
      }
    }
  }
}

template <int D, int WAVES>
__global__ __launch_bounds__(WAVES * WAVE_SIZE, 2) void flash_bwd_dq_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dq, int B, int Sq, int Skv, int Hq, int Hk, int q_start,
    float scale, bool causal) {
  static_assert(D == 128);
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_rows = smem;                 // 8 KiB ([32][256B] swz)
  char* v_rows = smem + 8 * 1024;      // 8 KiB
  char* kt = smem + 16 * 1024;         // 8 KiB ([128][64B] swz)
  char* tbuf = smem + 24 * 1024;       // per-wave 2 KiB x WAVES
  float* stats = reinterpret_cast<float*>(smem + 24 * 1024 + WAVES * 2048);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid / 64;
  const int col = lane & 31;
  const int half = lane >> 5;

  const int qb_blk = blockIdx.x * (WAVES * 32);
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (Hq / Hk);
  const int q0 = qb_blk + wid * 32;

  const long q_base = (((long)b * Sq) * Hq + h) * D;
  const long kv_base = (((long)b * Skv) * Hk + kvh) * D;
  const long q_rs = (long)Hq * D, kv_rs = (long)Hk * D;
  const float* lse_row = lse + ((long)b * Hq + h) * Sq;
  const float* dlt_row = delta + ((long)b * Hq + h) * Sq;

  // Q/dO rows of this wave as A-fragments
  bf16x8_v qfrag[D / 16], dofrag[D / 16];
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    const long off = q_base + (long)(q0 + col) * q_rs + c * 16 + half * 8;
    qfrag[c] = *reinterpret_cast<const bf16x8_v*>(q + off);
    dofrag[c] = *reinterpret_cast<const bf16x8_v*>(dout + off);
  }
  // lse/delta for the block's q rows -> LDS (32 regs saved per lane)
  if (tid < WAVES * 32) {
    stats[tid] = lse_row[qb_blk + tid];
    stats[WAVES * 32 + tid] = dlt_row[qb_blk + tid];
  }

  f32x16 dq_acc[D / 32];
#pragma unroll
  for (int t = 0; t < D / 32; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[t][r] = 0.f;

  char* tb = tbuf + wid * 2048;
  const int block_q_max = q_start + qb_blk + WAVES * 32 - 1;
  const int n_tiles = causal ? (min(block_q_max, Skv - 1) / 32 + 1) : (Skv + 31) / 32;

  for (int jk = 0; jk < n_tiles; ++jk) {
    const int k0 = jk * 32;
    // ---- stage K rows, V rows, K^T
    {
      const int c0 = (tid % 16) * 8;
      for (int rr = tid / 16; rr < 32; rr += WAVES * 4) {
        const int ks = min(k0 + rr, Skv - 1);  // overhang masked in compute
        bf16x8 kv8 = *reinterpret_cast<const bf16x8*>(k + kv_base + (long)ks * kv_rs + c0);
        bf16x8 vv8 = *reinterpret_cast<const bf16x8*>(v + kv_base + (long)ks * kv_rs + c0);
        *reinterpret_cast<bf16x8*>(k_rows + k_lds_off(rr, c0 * 2)) = kv8;
        *reinterpret_cast<bf16x8*>(v_rows + k_lds_off(rr, c0 * 2)) = vv8;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          *reinterpret_cast<bf16*>(kt + vt_lds_off(c0 + j, rr * 2)) = kv8.v[j];
      }
    }
    __syncthreads();

    const bool live = !causal || (k0 <= q_start + q0 + 31);
    if (live) {
      // ---- S[q][k] = Q K^T : A=Q rows, B from k_rows (contiguous d)
      f32x16 s;
#pragma unroll
      for (int r = 0; r < 16; ++r) s[r] = 0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        bf16x8_v kb = *reinterpret_cast<const bf16x8_v*>(
            k_rows + k_lds_off(col, (c * 16 + half * 8) * 2));
        s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfrag[c], kb, s, 0, 0, 0);
      }
      // ---- dP[q][k] = dO V^T
      f32x16 dp;
#pragma unroll
      for (int r = 0; r < 16; ++r) dp[r] = 0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        bf16x8_v vb = *reinterpret_cast<const bf16x8_v*>(
            v_rows + k_lds_off(col, (c * 16 + half * 8) * 2));
        dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dofrag[c], vb, dp, 0, 0, 0);
      }

      const int kg = k0 + col;   // D cols = k here
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;  // block-local + wid*32
        const int qg = q_start + q0 + qrow;
        const bool masked = (causal && kg > qg) || kg >= Skv;
        const float p = masked ? 0.f : __expf(s[r] * scale - stats[wid * 32 + qrow]);
        s[r] = p * (dp[r] - stats[WAVES * 32 + wid * 32 + qrow]) * scale;  // dS in place
      }

      // ---- transpose dS (cols=k -> A-frag rows=q), accumulate dQ
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
        *reinterpret_cast<bf16*>(tb + tb_off(qrow, col * 2)) = f2bf(s[r]);
      }
      __builtin_amdgcn_wave_barrier();
#pragma unroll
      for (int c2 = 0; c2 < 2; ++c2) {
        bf16x8_v da = *reinterpret_cast<const bf16x8_v*>(
            tb + tb_off(col, (c2 * 16 + half * 8) * 2));
#pragma unroll
        for (int t = 0; t < D / 32; ++t) {
          bf16x8_v ktb = *reinterpret_cast<const bf16x8_v*>(
              kt + vt_lds_off(t * 32 + col, (c2 * 16 + half * 8) * 2));
          dq_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, ktb, dq_acc[t], 0, 0, 0);
        }
      }
      __builtin_amdgcn_wave_barrier();
    }
    __syncthreads();
  }

  // ---- epilogue: plain bf16 stores (q rows exclusive to this block)
#pragma unroll
  for (int t = 0; t < D / 32; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = q0 + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (qrow < Sq) {
        dq[q_base + (long)qrow * q_rs + t * 32 + col] = f2bf(dq_acc[t][r]);
      }
    }
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> flash_attn_bwd(
    const at::Tensor& dout, const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& o, const at::Tensor& lse, double scale, bool causal,
    int64_t q_start) {
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1), Hk = k.size(2);
  TORCH_CHECK(D == 128 && Sq % 128 == 0 && Skv % 128 == 0,
              "flash_attn_bwd: D=128, Sq%128==0, Skv%128==0 required");
  auto stream = c10::hip::getCurrentHIPStream();

  auto delta = at::empty({B, Hq, Sq}, q.options().dtype(at::kFloat));
  {
    const long rows = (long)B * Sq * Hq;
    const int block = 256;
    const long grid = (rows * WAVE_SIZE + block - 1) / block;
    hipLaunchKernelGGL(fa_delta_kernel, dim3((unsigned)grid), dim3(block), 0, stream.stream(),
                       reinterpret_cast<const bf16*>(dout.data_ptr()),
                       reinterpret_cast<const bf16*>(o.data_ptr()),
                       delta.data_ptr<float>(), Sq, Hq, rows);
    HIP_CHECK_KERNEL();
  }

  auto dq = at::empty_like(q);
  auto dk32 = at::zeros({B, Skv, Hk, D}, q.options().dtype(at::kFloat));
  auto dv32 = at::zeros({B, Skv, Hk, D}, q.options().dtype(at::kFloat));

  if (Skv % 256 == 0) {
    const dim3 grid_kv(Skv / 256, Hq, B);
    const size_t smem_a = 32 * 1024 + 8 * 2048;
    hipLaunchKernelGGL((flash_bwd_dkv_kernel<128, 8>), grid_kv, dim3(512), smem_a,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dout.data_ptr()),
                       reinterpret_cast<const bf16*>(q.data_ptr()),
                       reinterpret_cast<const bf16*>(k.data_ptr()),
                       reinterpret_cast<const bf16*>(v.data_ptr()),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       dk32.data_ptr<float>(), dv32.data_ptr<float>(),
                       B, Sq, Skv, Hq, Hk, (int)q_start, (float)scale, causal);
  } else {
    const dim3 grid_kv(Skv / 128, Hq, B);
    const size_t smem_a = 32 * 1024 + 4 * 2048;
    hipLaunchKernelGGL((flash_bwd_dkv_kernel<128, 4>), grid_kv, dim3(256), smem_a,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dout.data_ptr()),
                       reinterpret_cast<const bf16*>(q.data_ptr()),
                       reinterpret_cast<const bf16*>(k.data_ptr()),
                       reinterpret_cast<const bf16*>(v.data_ptr()),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       dk32.data_ptr<float>(), dv32.data_ptr<float>(),
                       B, Sq, Skv, Hq, Hk, (int)q_start, (float)scale, causal);
  }
  HIP_CHECK_KERNEL();

  if (Sq % 256 == 0) {
    const dim3 grid_q(Sq / 256, Hq, B);
    const size_t smem_b = 24 * 1024 + 8 * 2048 + 8 * 64 * sizeof(float);
    hipLaunchKernelGGL((flash_bwd_dq_kernel<128, 8>), grid_q, dim3(512), smem_b,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dout.data_ptr()),
                       reinterpret_cast<const bf16*>(q.data_ptr()),
                       reinterpret_cast<const bf16*>(k.data_ptr()),
                       reinterpret_cast<const bf16*>(v.data_ptr()),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       reinterpret_cast<bf16*>(dq.data_ptr()),
                       B, Sq, Skv, Hq, Hk, (int)q_start, (float)scale, causal);
  } else {
    const dim3 grid_q(Sq / 128, Hq, B);
    const size_t smem_b = 24 * 1024 + 4 * 2048 + 4 * 64 * sizeof(float);
    hipLaunchKernelGGL((flash_bwd_dq_kernel<128, 4>), grid_q, dim3(256), smem_b,
                       stream.stream(),
                       reinterpret_cast<const bf16*>(dout.data_ptr()),
                       reinterpret_cast<const bf16*>(q.data_ptr()),
                       reinterpret_cast<const bf16*>(k.data_ptr()),
                       reinterpret_cast<const bf16*>(v.data_ptr()),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       reinterpret_cast<bf16*>(dq.data_ptr()),
                       B, Sq, Skv, Hq, Hk, (int)q_start, (float)scale, causal);
  }
  HIP_CHECK_KERNEL();

  return {dq, dk32.to(at::kBFloat16), dv32.to(at::kBFloat16)};
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
