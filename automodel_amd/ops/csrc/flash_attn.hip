// Flash attention fwd+bwd (CDNA4, gfx950) — BSHD + packed varlen (THD),
// causal, GQA, head dims {64, 96, 128, 192, 256} and split-dim MLA (qk 192 /
// v 128). Deterministic backward (no atomics).
//
// MI355X-native replacement for the reference's TE DotProductAttention /
// flash-attn backends (te_attention.py:352 arbitrary
// dims, distributed/thd_utils.py:85 THD varlen). Structure follows the CDNA4
// guide's fused-attention recipe (cdna_hip_programming.md Appendix B):
//   * swapped QK^T — mfma_f32_32x32x16_bf16 computing mfma(K, Q) so each
//     lane holds a P column for ONE q row -> softmax is in-register
//     (16 regs + one __shfl_xor(32) cross-half exchange)
//   * online softmax (running m, l per q row)
//   * P -> bf16 via packed cvt + __builtin_amdgcn_permlane32_swap to build
//     the PV A-fragment without LDS round trips (guide T12)
//   * K tile LDS-staged with XOR swizzle (guide T2 / Guideline 4), row
//     stride padded to a multiple of 256 B so the ((row&15)<<4) swizzle is
//     conflict-free at every head dim
//   * V tile staged TRANSPOSED in LDS (vt[d][k]) with its own XOR swizzle
//     so PV B-fragments are contiguous ds_read_b128
//
// Varlen (packed THD): per-token doc_start/doc_end arrays (precomputed from
// cu_seqlens in ops/attention.py) gate the causal mask to the query's own
// document; KV-tile ranges are clipped per block/wave so cross-document
// tiles are skipped, giving ONE kernel launch per packed batch.
//
// Backward determinism: the dKV kernel's grid is (kv tiles, Hk, B) and each
// block loops over the GQA query-head group, accumulating dK/dV in registers
// across the whole q loop — plain stores, no atomicAdd, bitwise
// run-to-run-identical grads (reference determinism discipline,
// moe/experts.py:66).

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

// LDS row-major tile offsets: row stride padded to a multiple of 256 B so
// every row starts at bank 0; ((row&15)<<4) rotates the 16-B slot per row ->
// a 16-lane ds_read_b128 group (16 distinct rows, same in-row byte) hits 16
// distinct slots (guide Guideline 4). PMC evidence: profiles/r1_pmc_*.csv.
__host__ __device__ constexpr int pad256(int bytes) { return ((bytes + 255) / 256) * 256; }

__device__ __forceinline__ int row_lds_off(int row, int byte_in_row, int stride_b) {
  return row * stride_b + (byte_in_row ^ ((row & 15) << 4));
}
// V^T-style tile: [D][KVBLK] bf16, 64 B rows (16 words). Rows d and d+4 share
// the same 16-word bank window; ((d>>2)^(d>>3))&3 rotation separates the four
// colliding rows for reads AND spreads the transpose-store writers.
__device__ __forceinline__ int vt_lds_off(int d, int byte_in_row) {
  return d * 64 + (byte_in_row ^ ((((d >> 2) ^ (d >> 3)) & 3) << 4));
}

// ---- gfx950 hardware-transpose (ds_read_b64_tr_b16) tile images -----------
// Probe-verified semantics (benchmarks/tr16_map.py, guide T10): each 16-lane
// group reads a contiguous 128-B region as a [4 row][16 col] bf16 block and
// lane g receives COLUMN g (one value per row). A [32 k][D] tile stored as
// [D/16 subtiles][32 k rows][16 d cols] therefore serves BOTH access
// patterns with no scalar transposes:
//   * row-chunk b128 reads (A-style: lane = k row, 8 consecutive d)
//   * transposed B-fragments via two tr16 reads (lane = d col, 8 k rows)
// Subtile stride 1152 B (not 1024): adjacent subtiles land 32 banks apart,
// so the two 16-lane halves of a wave (reading subtiles 2t and 2t+1) never
// share a bank window (b64 conflict groups are 32 lanes).
__device__ __forceinline__ int sub_off(int k, int d) {
  return (d >> 4) * 1152 + k * 32 + (d & 15) * 2;
}
template <int D>
__host__ __device__ constexpr int sub_img_bytes() { return (D / 16) * 1152; }

__device__ __forceinline__ unsigned lds_addr(const void* p) {
  return (unsigned)(unsigned long long)(uintptr_t)p;
}

// B-fragment for mfma_32x32x16: element j = T[k = 8*khalf + j][col =
// tile32*32 + (lane&31)] from a subtiled image, via two hardware-transpose
// reads (k rows 8h..8h+3 and 8h+4..8h+7). The trailing s_waitcnt rides in
// the second asm so the consuming mfma (which needs both halves) cannot be
// scheduled before the data lands.
// pipelined form: issue one fragment's two reads; wait-and-take later with
// a counted lgkmcnt (the "+v" operands order the consuming MFMAs after it)
__device__ __forceinline__ void tr16_issue2(unsigned addr, unsigned long long& r0,
                                            unsigned long long& r1) {
  asm volatile("ds_read_b64_tr_b16 %0, %2\n\tds_read_b64_tr_b16 %1, %2 offset:128"
               : "=v"(r0), "=v"(r1) : "v"(addr));
}
__device__ __forceinline__ bf16x8_v tr16_take(unsigned long long& r0,
                                              unsigned long long& r1, bool newer2) {
  if (newer2) {
    asm volatile("s_waitcnt lgkmcnt(2)" : "+v"(r0), "+v"(r1));
  } else {
    asm volatile("s_waitcnt lgkmcnt(0)" : "+v"(r0), "+v"(r1));
  }
  ulonglong2 u{r0, r1};
  return __builtin_bit_cast(bf16x8_v, u);
}

__device__ __forceinline__ bf16x8_v tr16_frag(unsigned img_base, int tile32,
                                              int khalf, int lane) {
  const unsigned a0 = img_base + (unsigned)((tile32 * 2 + ((lane >> 4) & 1)) * 1152 +
                                            (8 * khalf) * 32 + (lane & 15) * 8);
  unsigned long long v0, v1;
  // single asm block: outputs are fully landed when it retires, so the
  // register allocator can never spill an in-flight read (a real silent-
  // corruption hazard caught at (192,192) under register pressure)
  asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
               "ds_read_b64_tr_b16 %1, %2 offset:128\n\t"
               "s_waitcnt lgkmcnt(0)"
               : "=v"(v0), "=v"(v1) : "v"(a0));
  ulonglong2 u{v0, v1};
  return __builtin_bit_cast(bf16x8_v, u);
}

// Pipelining across other code is only safe when nothing spills: gate the
// pipelined tr16 paths to instantiations comfortably inside the register
// budget; the big-D variants take the self-contained tr16_frag instead.
__host__ __device__ constexpr bool pipe_ok_dkv(int DQK, int DV) {
  return 3 * (DQK + DV) / 4 + 16 <= 208;
}
__host__ __device__ constexpr bool pipe_ok_dq(int DQK, int DV) {
  return DQK / 4 + DV / 4 + DQK / 2 + 48 <= 208;
}
// per-wave transpose buffer: [32][64 B]
__device__ __forceinline__ int tb_off(int row, int byte_in_row) {
  return row * 64 + (byte_in_row ^ ((((row >> 2) ^ (row >> 3)) & 3) << 4));
}

// Min-waves-per-SIMD for the backward kernels: the f32 accumulators scale
// with head dim; past ~240 VGPR/lane two waves/SIMD would spill heavily, so
// the big-D instantiations run one wave/SIMD (4-wave workgroups).
__host__ __device__ constexpr int occ_dkv(int DQK, int DV) {
  return (3 * (DQK + DV) / 4 + 16) <= 240 ? 2 : 1;
}
__host__ __device__ constexpr int occ_dq(int DQK, int DV) {
  return (DQK / 4 + DV / 4 + DQK / 2 + 48) <= 240 ? 2 : 1;
}

// ===========================================================================
// Forward
// ===========================================================================

template <int DQK, int DV, int WAVES, bool VARLEN>
__global__ __launch_bounds__(WAVES * WAVE_SIZE, 2) void flash_fwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k, const bf16* __restrict__ v,
    bf16* __restrict__ out, float* __restrict__ lse, const int* __restrict__ doc_start,
    int B, int Sq, int Skv, int Hq, int Hk, int q_start, float scale, bool causal) {
  constexpr int KB = pad256(DQK * 2);     // K tile row stride (bytes)
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                               // KVBLK*KB
  char* v_img = smem + KVBLK * KB;                  // subtiled [DV/16][32][16]
  float* bcast = reinterpret_cast<float*>(smem + KVBLK * KB + sub_img_bytes<DV>());
  int* bcast_skip = reinterpret_cast<int*>(bcast + WAVES * 32);

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE_SIZE - 1);
  const int wid = tid / WAVE_SIZE;
  const int col = lane & 31;           // q row within wave tile (QK layout)
  const int half = lane >> 5;

  const int qblk0 = blockIdx.x * (WAVES * QBLK);
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (Hq / Hk);

  const long q_base = (((long)b * Sq) * Hq + h) * DQK;
  const long k_base = (((long)b * Skv) * Hk + kvh) * DQK;
  const long v_base = (((long)b * Skv) * Hk + kvh) * DV;
  const long o_base = (((long)b * Sq) * Hq + h) * DV;
  const long q_rs = (long)Hq * DQK, k_rs = (long)Hk * DQK;
  const long v_rs = (long)Hk * DV, o_rs = (long)Hq * DV;

  const int q0 = qblk0 + wid * QBLK;   // this wave's first q row

  // ---- load Q fragments to registers: lane holds Q[q0+col][8*half + j + 16*c]
  bf16x8_v qfrag[DQK / 16];
#pragma unroll
  for (int c = 0; c < DQK / 16; ++c) {
    const bf16* src = q + q_base + (long)(q0 + col) * q_rs + c * 16 + half * 8;
    qfrag[c] = *reinterpret_cast<const bf16x8_v*>(src);
  }

  // varlen doc bounds: lane's q row is fixed for the whole kernel
  const int qg_lane = q_start + q0 + col;
  const int ds_lane = VARLEN ? doc_start[qg_lane] : 0;        // lane's doc start
  const int ds_wave = VARLEN ? doc_start[q_start + q0] : 0;   // wave min (non-decreasing)

  // ---- accumulators
  f32x16 o_acc[DV / 32];
#pragma unroll
  for (int t = 0; t < DV / 32; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[t][r] = 0.f;
  float m_run = -1e30f, l_run = 0.f;

  const int q_block_max = q_start + qblk0 + WAVES * QBLK - 1;
  const int n_tiles = causal ? (min(q_block_max, Skv - 1) / KVBLK + 1)
                             : (Skv + KVBLK - 1) / KVBLK;
  // block-uniform start tile (varlen: nothing before the block's first doc)
  const int jt0 = VARLEN ? doc_start[q_start + qblk0] / KVBLK : 0;

  for (int jt = jt0; jt < n_tiles; ++jt) {
    const int k0 = jt * KVBLK;
    // ---- stage K tile (swizzled) and V^T tile cooperatively
    {
      constexpr int KCH = DQK / 8;     // 16-B chunks per K row
      for (int idx = tid; idx < KVBLK * KCH; idx += WAVES * WAVE_SIZE) {
        const int row = idx / KCH, c0 = (idx % KCH) * 8;
        const int ks = min(k0 + row, Skv - 1);  // overhang rows masked later
        *reinterpret_cast<bf16x8*>(k_lds + row_lds_off(row, c0 * 2, KB)) =
            *reinterpret_cast<const bf16x8*>(k + k_base + (long)ks * k_rs + c0);
      }
      constexpr int VCH = DV / 8;
      for (int idx = tid; idx < KVBLK * VCH; idx += WAVES * WAVE_SIZE) {
        const int row = idx / VCH, c0 = (idx % VCH) * 8;
        const int ks = min(k0 + row, Skv - 1);
        // vectorized subtile store (the round-1 path scatter-stored 8
        // scalars per vector to build a transposed image)
        *reinterpret_cast<bf16x8*>(v_img + sub_off(row, c0)) =
            *reinterpret_cast<const bf16x8*>(v + v_base + (long)ks * v_rs + c0);
      }
    }
    __syncthreads();

    const bool tile_live = (!causal || (k0 <= q_start + q0 + QBLK - 1)) &&
                           (!VARLEN || (k0 + KVBLK > ds_wave));
    float alpha = 1.f;
    bf16x8_v pa0, pa1;
    if (tile_live) {
      // ---- QK^T swapped: P[k][q] = sum_d K[k][d] * Q[q][d]
      f32x16 p;
#pragma unroll
      for (int r = 0; r < 16; ++r) p[r] = 0.f;
#pragma unroll
      for (int c = 0; c < DQK / 16; ++c) {
        bf16x8_v ka = *reinterpret_cast<const bf16x8_v*>(
            k_lds + row_lds_off(col, (c * 16 + half * 8) * 2, KB));
        p = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[c], p, 0, 0, 0);
      }

      // ---- masked online softmax, IN PLACE in the p accumulator
      const int qg = qg_lane;
      float tile_max = -1e30f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kg = k0 + (r & 3) + 8 * (r >> 2) + 4 * half;
        float sv = p[r] * scale;
        if ((causal && kg > qg) || kg >= Skv || (VARLEN && kg < ds_lane)) sv = -1e30f;
        p[r] = sv;
        tile_max = fmaxf(tile_max, sv);
      }
      tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32));
      // defer-max (guide T13, THR=8): if no row's max grew past m+THR, keep
      // the old max and skip the O-wide rescale pass — P is then bounded by
      // e^8 instead of 1, which the f32 accumulator tolerates. Wave-uniform
      // decision taken BEFORE this tile's P is exponentiated (the textbook-
      // safe order; guide T13 correctness hazard).
      const bool need_rescale = !__all(tile_max <= m_run + 8.0f);
      const float m_new = need_rescale ? fmaxf(m_run, tile_max) : m_run;
      alpha = need_rescale ? __expf(m_run - m_new) : 1.f;
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
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        auto r02 = __builtin_amdgcn_permlane32_swap(pk[4 * i + 0], pk[4 * i + 2], false, false);
        auto r13 = __builtin_amdgcn_permlane32_swap(pk[4 * i + 1], pk[4 * i + 3], false, false);
        unsigned frag[4] = {(unsigned)r02[0], (unsigned)r13[0], (unsigned)r02[1], (unsigned)r13[1]};
        if (i == 0) pa0 = __builtin_bit_cast(bf16x8_v, *reinterpret_cast<ulonglong2*>(frag));
        else pa1 = __builtin_bit_cast(bf16x8_v, *reinterpret_cast<ulonglong2*>(frag));
      }

      // ---- broadcast alpha to O layout via LDS (sign bit of slot 0 set
      // when the whole wave deferred, so the rescale pass can be skipped)
      if (lane < 32) bcast[wid * 32 + lane] = alpha;
      if (lane == 0) bcast_skip[wid] = need_rescale ? 1 : 0;
    }
    // (waves with dead tiles skip compute but still hit the barriers)
    __syncthreads();

    if (tile_live) {
      // ---- rescale O by alpha (skipped on deferred tiles, guide T13)
      if (bcast_skip[wid]) {
#pragma unroll
        for (int t = 0; t < DV / 32; ++t) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int row = (r & 3) + 8 * (r >> 2) + 4 * half;
            o_acc[t][r] *= bcast[wid * 32 + row];
          }
        }
      }
      // software-pipelined tr16 PV: tile t+1's four transpose reads are in
      // flight during tile t's MFMAs; the counted s_waitcnt takes the
      // fragment registers as "+v" operands so the consuming MFMAs cannot
      // be scheduled past it. lgkmcnt(4) is conservative-safe: compiler-
      // issued LDS ops between ours only strengthen the wait.
      const unsigned v_base_a = lds_addr(v_img);
      if constexpr (DV > 128) {
#pragma unroll
        for (int t = 0; t < DV / 32; ++t) {
          bf16x8_v vb0 = tr16_frag(v_base_a, t, half, lane);
          bf16x8_v vb1 = tr16_frag(v_base_a, t, half + 2, lane);
          o_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa0, vb0, o_acc[t], 0, 0, 0);
          o_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa1, vb1, o_acc[t], 0, 0, 0);
        }
      } else {
      unsigned long long f0a, f0b, f1a, f1b, g0a, g0b, g1a, g1b;
      auto tr_issue = [&](int t, unsigned long long& r0a, unsigned long long& r0b,
                          unsigned long long& r1a, unsigned long long& r1b) {
        const unsigned base = v_base_a +
            (unsigned)((t * 2 + ((lane >> 4) & 1)) * 1152 + (lane & 15) * 8);
        const unsigned klo = (unsigned)(8 * half) * 32;
        const unsigned khi = (unsigned)(8 * (half + 2)) * 32;
        asm volatile("ds_read_b64_tr_b16 %0, %4\n\t"
                     "ds_read_b64_tr_b16 %1, %4 offset:128\n\t"
                     "ds_read_b64_tr_b16 %2, %5\n\t"
                     "ds_read_b64_tr_b16 %3, %5 offset:128"
                     : "=v"(r0a), "=v"(r0b), "=v"(r1a), "=v"(r1b)
                     : "v"(base + klo), "v"(base + khi));
      };
      tr_issue(0, f0a, f0b, f1a, f1b);
#pragma unroll
      for (int t = 0; t < DV / 32; ++t) {
        unsigned long long& c0a = (t & 1) ? g0a : f0a;
        unsigned long long& c0b = (t & 1) ? g0b : f0b;
        unsigned long long& c1a = (t & 1) ? g1a : f1a;
        unsigned long long& c1b = (t & 1) ? g1b : f1b;
        if (t + 1 < DV / 32) {
          tr_issue(t + 1, (t & 1) ? f0a : g0a, (t & 1) ? f0b : g0b,
                   (t & 1) ? f1a : g1a, (t & 1) ? f1b : g1b);
          asm volatile("s_waitcnt lgkmcnt(4)"
                       : "+v"(c0a), "+v"(c0b), "+v"(c1a), "+v"(c1b));
        } else {
          asm volatile("s_waitcnt lgkmcnt(0)"
                       : "+v"(c0a), "+v"(c0b), "+v"(c1a), "+v"(c1b));
        }
        ulonglong2 u0{c0a, c0b}, u1{c1a, c1b};
        bf16x8_v vb0 = __builtin_bit_cast(bf16x8_v, u0);
        bf16x8_v vb1 = __builtin_bit_cast(bf16x8_v, u1);
        o_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa0, vb0, o_acc[t], 0, 0, 0);
        o_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa1, vb1, o_acc[t], 0, 0, 0);
      }
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
  for (int t = 0; t < DV / 32; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * half;  // q row in wave tile
      const int qrow = q0 + row;
      if (qrow < Sq) {
        const float inv_l = bcast[wid * 32 + row];
        out[o_base + (long)qrow * o_rs + t * 32 + col] = f2bf(o_acc[t][r] * inv_l);
      }
    }
  }
}

template <int DQK, int DV, int WAVES, bool VARLEN>
static void launch_fwd(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
                       at::Tensor& out, at::Tensor& lse, const int* doc_start,
                       int B, int Sq, int Skv, int Hq, int Hk, int q_start,
                       float scale, bool causal, hipStream_t stream) {
  const dim3 grid(Sq / (WAVES * QBLK), Hq, B);
  const size_t smem = KVBLK * pad256(DQK * 2) + sub_img_bytes<DV>() +
                      WAVES * 33 * sizeof(float);
  hipLaunchKernelGGL((flash_fwd_kernel<DQK, DV, WAVES, VARLEN>), grid,
                     dim3(WAVES * WAVE_SIZE), smem, stream,
                     reinterpret_cast<const bf16*>(q.data_ptr()),
                     reinterpret_cast<const bf16*>(k.data_ptr()),
                     reinterpret_cast<const bf16*>(v.data_ptr()),
                     reinterpret_cast<bf16*>(out.data_ptr()), lse.data_ptr<float>(),
                     doc_start, B, Sq, Skv, Hq, Hk, q_start, scale, causal);
  HIP_CHECK_KERNEL();
}

std::tuple<at::Tensor, at::Tensor> flash_attn_fwd(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v, double scale,
    bool causal, int64_t q_start, const std::optional<at::Tensor>& doc_start) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.scalar_type() == at::kBFloat16,
              "flash_attn_fwd: q must be [B,S,Hq,D] bf16");
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), Dqk = q.size(3);
  const int Skv = k.size(1), Hk = k.size(2), Dv = v.size(3);
  TORCH_CHECK(Hq % Hk == 0, "flash_attn_fwd: Hq must be divisible by Hk");
  TORCH_CHECK(Sq % (4 * QBLK) == 0, "flash_attn_fwd: Sq must be a multiple of 128");
  TORCH_CHECK(Skv % KVBLK == 0, "flash_attn_fwd: Skv must be a multiple of 32");
  const bool varlen = doc_start.has_value();
  const int* ds_ptr = varlen ? doc_start->data_ptr<int>() : nullptr;
  if (varlen) TORCH_CHECK(causal && B == 1, "varlen flash is causal with B==1");
  auto out = at::empty({B, Sq, Hq, Dv}, q.options());
  auto lse = at::empty({B, Hq, Sq}, q.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream().stream();

  bool done = false;
#define FA_FWD_CASE(A, C)                                                             \
  if (!done && Dqk == A && Dv == C) {                                                 \
    done = true;                                                                      \
    if (varlen) {                                                                     \
      launch_fwd<A, C, 4, true>(q, k, v, out, lse, ds_ptr, B, Sq, Skv, Hq, Hk,        \
                                (int)q_start, (float)scale, causal, stream);          \
    } else if (Sq % (8 * QBLK) == 0) {                                                \
      launch_fwd<A, C, 8, false>(q, k, v, out, lse, nullptr, B, Sq, Skv, Hq, Hk,      \
                                 (int)q_start, (float)scale, causal, stream);         \
    } else {                                                                          \
      launch_fwd<A, C, 4, false>(q, k, v, out, lse, nullptr, B, Sq, Skv, Hq, Hk,      \
                                 (int)q_start, (float)scale, causal, stream);         \
    }                                                                                 \
  }
  FA_FWD_CASE(64, 64)
  FA_FWD_CASE(96, 96)
  FA_FWD_CASE(128, 128)
  FA_FWD_CASE(192, 128)
  FA_FWD_CASE(192, 192)
  FA_FWD_CASE(256, 256)
#undef FA_FWD_CASE
  TORCH_CHECK(done, "flash_attn_fwd: unsupported head dims (Dqk=", Dqk, ", Dv=", Dv,
              "); supported pairs: (64,64),(96,96),(128,128),(192,128),(192,192),(256,256)");
  return {out, lse};
}

// ===========================================================================
// Backward: flash-attention-2 style split.
//   delta kernel: delta[b,h,s] = rowsum(dO * O)
//   kernel A (kv-parallel, DETERMINISTIC): grid over (kv tiles, Hk, B); each
//     block loops the GQA query-head group and all q tiles, accumulating
//     dK/dV in registers; plain bf16 stores (no atomics -> bitwise
//     reproducible grads).
//   kernel B (q-parallel): recompute S in [q][k] layout, accumulate dQ.
// ===========================================================================

__global__ void fa_delta_kernel(const bf16* __restrict__ dout, const bf16* __restrict__ o,
                                float* __restrict__ delta, int S, int Hq, int DV,
                                long rows) {
  // row r = ((b*S + s)*Hq + h); one wave per row
  const long row = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE_SIZE;
  if (row >= rows) return;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const bf16* dp = dout + row * DV;
  const bf16* op = o + row * DV;
  float acc = 0.f;
  for (int j = lane * 2; j < DV; j += 2 * WAVE_SIZE) {
    acc += bf2f(dp[j]) * bf2f(op[j]) + bf2f(dp[j + 1]) * bf2f(op[j + 1]);
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) {
    const long h = row % Hq;
    const long bs = row / Hq;
    const long b = bs / S, s = bs % S;
    delta[(b * Hq + h) * S + s] = acc;
  }
}

template <int DQK, int DV, int WAVES, bool VARLEN>
__global__ __launch_bounds__(WAVES * WAVE_SIZE, occ_dkv(DQK, DV)) void flash_bwd_dkv_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dk, bf16* __restrict__ dv, const int* __restrict__ doc_end,
    int B, int Sq, int Skv, int Hq, int Hk, int q_start, float scale, bool causal) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* q_img = smem;                                    // subtiled [DQK/16][32][16]
  char* do_img = smem + sub_img_bytes<DQK>();            // subtiled [DV/16][32][16]
  char* tbuf = do_img + sub_img_bytes<DV>();             // WAVES*2048

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid / 64;
  const int col = lane & 31;
  const int half = lane >> 5;

  const int kvb = blockIdx.x * (WAVES * 32);
  const int kvh = blockIdx.y;
  const int b = blockIdx.z;
  const int G = Hq / Hk;               // GQA group size
  const int kv0 = kvb + wid * 32;      // this wave's kv rows

  const long k_base = (((long)b * Skv) * Hk + kvh) * DQK;
  const long v_base = (((long)b * Skv) * Hk + kvh) * DV;
  const long k_rs = (long)Hk * DQK, v_rs = (long)Hk * DV;
  const long q_rs = (long)Hq * DQK, do_rs = (long)Hq * DV;

  // K/V rows of this wave -> A fragments in registers
  bf16x8_v kfrag[DQK / 16], vfrag[DV / 16];
#pragma unroll
  for (int c = 0; c < DQK / 16; ++c)
    kfrag[c] = *reinterpret_cast<const bf16x8_v*>(
        k + k_base + (long)(kv0 + col) * k_rs + c * 16 + half * 8);
#pragma unroll
  for (int c = 0; c < DV / 16; ++c)
    vfrag[c] = *reinterpret_cast<const bf16x8_v*>(
        v + v_base + (long)(kv0 + col) * v_rs + c * 16 + half * 8);

  // varlen: per-element doc_end for this wave's 16 kv rows (fixed all kernel)
  int dend[16];
  int dend_wave = Sq;                  // max over the wave's rows
  if (VARLEN) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kg = kv0 + (r & 3) + 8 * (r >> 2) + 4 * half;
      dend[r] = doc_end[min(kg, Skv - 1)];
    }
    dend_wave = doc_end[min(kv0 + 31, Skv - 1)];
  }

  f32x16 dk_acc[DQK / 32], dv_acc[DV / 32];
#pragma unroll
  for (int t = 0; t < DQK / 32; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dk_acc[t][r] = 0.f;
#pragma unroll
  for (int t = 0; t < DV / 32; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dv_acc[t][r] = 0.f;

  char* tb = tbuf + wid * 2048;
  const int jq_start = causal ? (kvb > q_start ? (kvb - q_start) / 32 : 0) : 0;
  // block-uniform end (varlen: nothing after the block's last doc)
  int jq_end = Sq / 32;
  if (VARLEN) {
    const int de_blk = doc_end[min(kvb + WAVES * 32 - 1, Skv - 1)];
    jq_end = min(jq_end, (de_blk - q_start + 31) / 32);
  }

  for (int g = 0; g < G; ++g) {
    const int h = kvh * G + g;
    const long q_base = (((long)b * Sq) * Hq + h) * DQK;
    const long do_base = (((long)b * Sq) * Hq + h) * DV;
    const float* lse_row = lse + ((long)b * Hq + h) * Sq;
    const float* dlt_row = delta + ((long)b * Hq + h) * Sq;

    for (int jq = jq_start; jq < jq_end; ++jq) {
      const int q0 = jq * 32;
      // ---- cooperative stage: ONE subtiled image per tensor (serves both
      // the row-chunk B-frags and the tr16 transposed B-frags — the round-1
      // kernel kept a row image AND a scalar-scattered transposed image)
      {
        constexpr int QCH = DQK / 8;
        for (int idx = tid; idx < 32 * QCH; idx += WAVES * 64) {
          const int rr = idx / QCH, c0 = (idx % QCH) * 8;
          *reinterpret_cast<bf16x8*>(q_img + sub_off(rr, c0)) =
              *reinterpret_cast<const bf16x8*>(q + q_base + (long)(q0 + rr) * q_rs + c0);
        }
        constexpr int OCH = DV / 8;
        for (int idx = tid; idx < 32 * OCH; idx += WAVES * 64) {
          const int rr = idx / OCH, c0 = (idx % OCH) * 8;
          *reinterpret_cast<bf16x8*>(do_img + sub_off(rr, c0)) =
              *reinterpret_cast<const bf16x8*>(
                  dout + do_base + (long)(q0 + rr) * do_rs + c0);
        }
      }
      __syncthreads();

      const bool live = (!causal || (q_start + q0 + 31 >= kv0)) &&
                        (!VARLEN || (q_start + q0 < dend_wave));
      if (live) {
        const int ql = q0 + col;          // local q row (lse/delta index)
        const int qg = q_start + ql;      // global position (mask)
        const float lse_q = lse_row[ql];
        const float dlt_q = dlt_row[ql];

        // ---- S^T = K Q^T (rows = kv, cols = q)
        f32x16 st;
#pragma unroll
        for (int r = 0; r < 16; ++r) st[r] = 0.f;
#pragma unroll
        for (int c = 0; c < DQK / 16; ++c) {
          bf16x8_v qb = *reinterpret_cast<const bf16x8_v*>(
              q_img + sub_off(col, c * 16 + half * 8));
          st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag[c], qb, st, 0, 0, 0);
        }
        // ---- dP^T = V dO^T
        f32x16 dpt;
#pragma unroll
        for (int r = 0; r < 16; ++r) dpt[r] = 0.f;
#pragma unroll
        for (int c = 0; c < DV / 16; ++c) {
          bf16x8_v db = *reinterpret_cast<const bf16x8_v*>(
              do_img + sub_off(col, c * 16 + half * 8));
          dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag[c], db, dpt, 0, 0, 0);
        }

        // in place: st becomes P^T, dpt becomes dS^T
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kg = kv0 + (r & 3) + 8 * (r >> 2) + 4 * half;
          const bool masked = (causal && kg > qg) || kg >= Skv ||
                              (VARLEN && qg >= dend[r]);
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
        const unsigned do_base_a = lds_addr(do_img);
        if constexpr (!pipe_ok_dkv(DQK, DV)) {
#pragma unroll
          for (int c2 = 0; c2 < 2; ++c2) {
            bf16x8_v pa = *reinterpret_cast<const bf16x8_v*>(
                tb + tb_off(col, (c2 * 16 + half * 8) * 2));
#pragma unroll
            for (int t = 0; t < DV / 32; ++t) {
              bf16x8_v dob = tr16_frag(do_base_a, t, c2 * 2 + half, lane);
              dv_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, dob, dv_acc[t], 0, 0, 0);
            }
          }
        } else {
          bf16x8_v paf[2];
#pragma unroll
          for (int c2 = 0; c2 < 2; ++c2)
            paf[c2] = *reinterpret_cast<const bf16x8_v*>(
                tb + tb_off(col, (c2 * 16 + half * 8) * 2));
          constexpr int NT = DV / 32;
          unsigned long long pA, pB, qA, qB;
          auto addr_f = [&](int i) {
            const int c2 = i / NT, t = i % NT;
            return do_base_a + (unsigned)((t * 2 + ((lane >> 4) & 1)) * 1152 +
                                          (8 * (c2 * 2 + half)) * 32 + (lane & 15) * 8);
          };
          tr16_issue2(addr_f(0), pA, pB);
#pragma unroll
          for (int i = 0; i < 2 * NT; ++i) {
            const bool last = (i + 1 == 2 * NT);
            if (!last) tr16_issue2(addr_f(i + 1), (i & 1) ? pA : qA, (i & 1) ? pB : qB);
            bf16x8_v dob = tr16_take((i & 1) ? qA : pA, (i & 1) ? qB : pB, !last);
            dv_acc[i % NT] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                paf[i / NT], dob, dv_acc[i % NT], 0, 0, 0);
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
        const unsigned q_base_a = lds_addr(q_img);
        if constexpr (!pipe_ok_dkv(DQK, DV)) {
#pragma unroll
          for (int c2 = 0; c2 < 2; ++c2) {
            bf16x8_v da = *reinterpret_cast<const bf16x8_v*>(
                tb + tb_off(col, (c2 * 16 + half * 8) * 2));
#pragma unroll
            for (int t = 0; t < DQK / 32; ++t) {
              bf16x8_v qb2 = tr16_frag(q_base_a, t, c2 * 2 + half, lane);
              dk_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, qb2, dk_acc[t], 0, 0, 0);
            }
          }
        } else {
          bf16x8_v daf[2];
#pragma unroll
          for (int c2 = 0; c2 < 2; ++c2)
            daf[c2] = *reinterpret_cast<const bf16x8_v*>(
                tb + tb_off(col, (c2 * 16 + half * 8) * 2));
          constexpr int NT = DQK / 32;
          unsigned long long pA, pB, qA, qB;
          auto addr_f = [&](int i) {
            const int c2 = i / NT, t = i % NT;
            return q_base_a + (unsigned)((t * 2 + ((lane >> 4) & 1)) * 1152 +
                                         (8 * (c2 * 2 + half)) * 32 + (lane & 15) * 8);
          };
          tr16_issue2(addr_f(0), pA, pB);
#pragma unroll
          for (int i = 0; i < 2 * NT; ++i) {
            const bool last = (i + 1 == 2 * NT);
            if (!last) tr16_issue2(addr_f(i + 1), (i & 1) ? pA : qA, (i & 1) ? pB : qB);
            bf16x8_v qb2 = tr16_take((i & 1) ? qA : pA, (i & 1) ? qB : pB, !last);
            dk_acc[i % NT] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                daf[i / NT], qb2, dk_acc[i % NT], 0, 0, 0);
          }
        }
      }
      __syncthreads();
    }
  }

  // ---- epilogue: plain bf16 stores — each kv row is owned by exactly one
  // wave of one block (grid is per-Hk), so no cross-block accumulation.
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int krow = (r & 3) + 8 * (r >> 2) + 4 * half;
    const int kg = kv0 + krow;
    if (kg < Skv) {
#pragma unroll
      for (int t = 0; t < DQK / 32; ++t)
        dk[k_base + (long)kg * k_rs + t * 32 + col] = f2bf(dk_acc[t][r]);
#pragma unroll
      for (int t = 0; t < DV / 32; ++t)
        dv[v_base + (long)kg * v_rs + t * 32 + col] = f2bf(dv_acc[t][r]);
    }
  }
}

template <int DQK, int DV, int WAVES, bool VARLEN>
__global__ __launch_bounds__(WAVES * WAVE_SIZE, occ_dq(DQK, DV)) void flash_bwd_dq_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dq, const int* __restrict__ doc_start, int B, int Sq, int Skv,
    int Hq, int Hk, int q_start, float scale, bool causal) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_img = smem;                                  // subtiled [DQK/16][32][16]
  char* v_img = smem + sub_img_bytes<DQK>();           // subtiled [DV/16][32][16]
  char* tbuf = v_img + sub_img_bytes<DV>();            // WAVES*2048
  float* stats = reinterpret_cast<float*>(tbuf + WAVES * 2048);  // 3*WAVES*32

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

  const long q_base = (((long)b * Sq) * Hq + h) * DQK;
  const long do_base = (((long)b * Sq) * Hq + h) * DV;
  const long k_base = (((long)b * Skv) * Hk + kvh) * DQK;
  const long v_base = (((long)b * Skv) * Hk + kvh) * DV;
  const long q_rs = (long)Hq * DQK, do_rs = (long)Hq * DV;
  const long k_rs = (long)Hk * DQK, v_rs = (long)Hk * DV;
  const float* lse_row = lse + ((long)b * Hq + h) * Sq;
  const float* dlt_row = delta + ((long)b * Hq + h) * Sq;

  // Q/dO rows of this wave as A-fragments
  bf16x8_v qfrag[DQK / 16], dofrag[DV / 16];
#pragma unroll
  for (int c = 0; c < DQK / 16; ++c)
    qfrag[c] = *reinterpret_cast<const bf16x8_v*>(
        q + q_base + (long)(q0 + col) * q_rs + c * 16 + half * 8);
#pragma unroll
  for (int c = 0; c < DV / 16; ++c)
    dofrag[c] = *reinterpret_cast<const bf16x8_v*>(
        dout + do_base + (long)(q0 + col) * do_rs + c * 16 + half * 8);
  // lse/delta(/doc_start) for the block's q rows -> LDS
  if (tid < WAVES * 32) {
    stats[tid] = lse_row[qb_blk + tid];
    stats[WAVES * 32 + tid] = dlt_row[qb_blk + tid];
    if (VARLEN) stats[2 * WAVES * 32 + tid] = (float)doc_start[q_start + qb_blk + tid];
  }

  f32x16 dq_acc[DQK / 32];
#pragma unroll
  for (int t = 0; t < DQK / 32; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[t][r] = 0.f;

  char* tb = tbuf + wid * 2048;
  const int ds_wave = VARLEN ? doc_start[q_start + q0] : 0;
  const int block_q_max = q_start + qb_blk + WAVES * 32 - 1;
  const int n_tiles = causal ? (min(block_q_max, Skv - 1) / 32 + 1) : (Skv + 31) / 32;
  const int jk0 = VARLEN ? doc_start[q_start + qb_blk] / 32 : 0;

  for (int jk = jk0; jk < n_tiles; ++jk) {
    const int k0 = jk * 32;
    // ---- stage K and V as subtiled images (one image serves both the
    // row-chunk and the tr16 transposed reads)
    {
      constexpr int KCH = DQK / 8;
      for (int idx = tid; idx < 32 * KCH; idx += WAVES * 64) {
        const int rr = idx / KCH, c0 = (idx % KCH) * 8;
        const int ks = min(k0 + rr, Skv - 1);  // overhang masked in compute
        *reinterpret_cast<bf16x8*>(k_img + sub_off(rr, c0)) =
            *reinterpret_cast<const bf16x8*>(k + k_base + (long)ks * k_rs + c0);
      }
      constexpr int VCH = DV / 8;
      for (int idx = tid; idx < 32 * VCH; idx += WAVES * 64) {
        const int rr = idx / VCH, c0 = (idx % VCH) * 8;
        const int ks = min(k0 + rr, Skv - 1);
        *reinterpret_cast<bf16x8*>(v_img + sub_off(rr, c0)) =
            *reinterpret_cast<const bf16x8*>(v + v_base + (long)ks * v_rs + c0);
      }
    }
    __syncthreads();

    const bool live = (!causal || (k0 <= q_start + q0 + 31)) &&
                      (!VARLEN || (k0 + 31 >= ds_wave));
    if (live) {
      // ---- S[q][k] = Q K^T : A=Q rows, B from k_rows (contiguous d)
      f32x16 s;
#pragma unroll
      for (int r = 0; r < 16; ++r) s[r] = 0.f;
#pragma unroll
      for (int c = 0; c < DQK / 16; ++c) {
        bf16x8_v kb = *reinterpret_cast<const bf16x8_v*>(
            k_img + sub_off(col, c * 16 + half * 8));
        s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfrag[c], kb, s, 0, 0, 0);
      }
      // ---- dP[q][k] = dO V^T
      f32x16 dp;
#pragma unroll
      for (int r = 0; r < 16; ++r) dp[r] = 0.f;
#pragma unroll
      for (int c = 0; c < DV / 16; ++c) {
        bf16x8_v vb = *reinterpret_cast<const bf16x8_v*>(
            v_img + sub_off(col, c * 16 + half * 8));
        dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dofrag[c], vb, dp, 0, 0, 0);
      }

      const int kg = k0 + col;   // cols = k here
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;  // block-local + wid*32
        const int qg = q_start + q0 + qrow;
        const bool masked = (causal && kg > qg) || kg >= Skv ||
                            (VARLEN && kg < (int)stats[2 * WAVES * 32 + wid * 32 + qrow]);
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
      const unsigned k_base_a = lds_addr(k_img);
      if constexpr (!pipe_ok_dq(DQK, DV)) {
#pragma unroll
        for (int c2 = 0; c2 < 2; ++c2) {
          bf16x8_v da = *reinterpret_cast<const bf16x8_v*>(
              tb + tb_off(col, (c2 * 16 + half * 8) * 2));
#pragma unroll
          for (int t = 0; t < DQK / 32; ++t) {
            bf16x8_v ktb = tr16_frag(k_base_a, t, c2 * 2 + half, lane);
            dq_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, ktb, dq_acc[t], 0, 0, 0);
          }
        }
      } else {
        bf16x8_v daf[2];
#pragma unroll
        for (int c2 = 0; c2 < 2; ++c2)
          daf[c2] = *reinterpret_cast<const bf16x8_v*>(
              tb + tb_off(col, (c2 * 16 + half * 8) * 2));
        constexpr int NT = DQK / 32;
        unsigned long long pA, pB, qA, qB;
        auto addr_f = [&](int i) {
          const int c2 = i / NT, t = i % NT;
          return k_base_a + (unsigned)((t * 2 + ((lane >> 4) & 1)) * 1152 +
                                       (8 * (c2 * 2 + half)) * 32 + (lane & 15) * 8);
        };
        tr16_issue2(addr_f(0), pA, pB);
#pragma unroll
        for (int i = 0; i < 2 * NT; ++i) {
          const bool last = (i + 1 == 2 * NT);
          if (!last) tr16_issue2(addr_f(i + 1), (i & 1) ? pA : qA, (i & 1) ? pB : qB);
          bf16x8_v ktb = tr16_take((i & 1) ? qA : pA, (i & 1) ? qB : pB, !last);
          dq_acc[i % NT] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              daf[i / NT], ktb, dq_acc[i % NT], 0, 0, 0);
        }
      }
      __builtin_amdgcn_wave_barrier();
    }
    __syncthreads();
  }

  // ---- epilogue: plain bf16 stores (q rows exclusive to this block)
#pragma unroll
  for (int t = 0; t < DQK / 32; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = q0 + (r & 3) + 8 * (r >> 2) + 4 * half;
      if (qrow < Sq) {
        dq[q_base + (long)qrow * q_rs + t * 32 + col] = f2bf(dq_acc[t][r]);
      }
    }
  }
}

template <int DQK, int DV, int WAVES, bool VARLEN>
static void launch_dkv(const at::Tensor& dout, const at::Tensor& q, const at::Tensor& k,
                       const at::Tensor& v, const at::Tensor& lse, const at::Tensor& delta,
                       at::Tensor& dk, at::Tensor& dv, const int* doc_end, int B, int Sq,
                       int Skv, int Hq, int Hk, int q_start, float scale, bool causal,
                       hipStream_t stream) {
  const dim3 grid(Skv / (WAVES * 32), Hk, B);
  const size_t smem = sub_img_bytes<DQK>() + sub_img_bytes<DV>() + WAVES * 2048;
  hipLaunchKernelGGL((flash_bwd_dkv_kernel<DQK, DV, WAVES, VARLEN>), grid,
                     dim3(WAVES * WAVE_SIZE), smem, stream,
                     reinterpret_cast<const bf16*>(dout.data_ptr()),
                     reinterpret_cast<const bf16*>(q.data_ptr()),
                     reinterpret_cast<const bf16*>(k.data_ptr()),
                     reinterpret_cast<const bf16*>(v.data_ptr()),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     reinterpret_cast<bf16*>(dk.data_ptr()),
                     reinterpret_cast<bf16*>(dv.data_ptr()), doc_end,
                     B, Sq, Skv, Hq, Hk, q_start, scale, causal);
  HIP_CHECK_KERNEL();
}

template <int DQK, int DV, int WAVES, bool VARLEN>
static void launch_dq(const at::Tensor& dout, const at::Tensor& q, const at::Tensor& k,
                      const at::Tensor& v, const at::Tensor& lse, const at::Tensor& delta,
                      at::Tensor& dq, const int* doc_start, int B, int Sq, int Skv, int Hq,
                      int Hk, int q_start, float scale, bool causal, hipStream_t stream) {
  const dim3 grid(Sq / (WAVES * 32), Hq, B);
  const size_t smem = sub_img_bytes<DQK>() + sub_img_bytes<DV>() +
                      WAVES * 2048 + 3 * WAVES * 32 * sizeof(float);
  hipLaunchKernelGGL((flash_bwd_dq_kernel<DQK, DV, WAVES, VARLEN>), grid,
                     dim3(WAVES * WAVE_SIZE), smem, stream,
                     reinterpret_cast<const bf16*>(dout.data_ptr()),
                     reinterpret_cast<const bf16*>(q.data_ptr()),
                     reinterpret_cast<const bf16*>(k.data_ptr()),
                     reinterpret_cast<const bf16*>(v.data_ptr()),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     reinterpret_cast<bf16*>(dq.data_ptr()), doc_start,
                     B, Sq, Skv, Hq, Hk, q_start, scale, causal);
  HIP_CHECK_KERNEL();
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> flash_attn_bwd(
    const at::Tensor& dout, const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& o, const at::Tensor& lse, double scale, bool causal, int64_t q_start,
    const std::optional<at::Tensor>& doc_start, const std::optional<at::Tensor>& doc_end) {
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), Dqk = q.size(3);
  const int Skv = k.size(1), Hk = k.size(2), Dv = v.size(3);
  TORCH_CHECK(Sq % 128 == 0 && Skv % 128 == 0,
              "flash_attn_bwd: Sq%128==0, Skv%128==0 required");
  const bool varlen = doc_start.has_value();
  TORCH_CHECK(varlen == doc_end.has_value(), "doc_start and doc_end go together");
  const int* ds_ptr = varlen ? doc_start->data_ptr<int>() : nullptr;
  const int* de_ptr = varlen ? doc_end->data_ptr<int>() : nullptr;
  auto stream = c10::hip::getCurrentHIPStream().stream();

  auto delta = at::empty({B, Hq, Sq}, q.options().dtype(at::kFloat));
  {
    const long rows = (long)B * Sq * Hq;
    const int block = 256;
    const long grid = (rows * WAVE_SIZE + block - 1) / block;
    hipLaunchKernelGGL(fa_delta_kernel, dim3((unsigned)grid), dim3(block), 0, stream,
                       reinterpret_cast<const bf16*>(dout.data_ptr()),
                       reinterpret_cast<const bf16*>(o.data_ptr()),
                       delta.data_ptr<float>(), Sq, Hq, Dv, rows);
    HIP_CHECK_KERNEL();
  }

  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);

  bool done = false;
  // kv-parallel dKV: 8-wave only where two waves/SIMD fit (occ_dkv == 2)
#define FA_BWD_CASE(A, C)                                                                \
  if (!done && Dqk == A && Dv == C) {                                                    \
    done = true;                                                                         \
    if (varlen) {                                                                        \
      launch_dkv<A, C, 4, true>(dout, q, k, v, lse, delta, dk, dv, de_ptr, B, Sq, Skv,   \
                                Hq, Hk, (int)q_start, (float)scale, causal, stream);     \
      launch_dq<A, C, 4, true>(dout, q, k, v, lse, delta, dq, ds_ptr, B, Sq, Skv, Hq,    \
                               Hk, (int)q_start, (float)scale, causal, stream);          \
    } else {                                                                             \
      if (occ_dkv(A, C) == 2 && Skv % 256 == 0) {                                        \
        launch_dkv<A, C, (occ_dkv(A, C) == 2 ? 8 : 4), false>(                           \
            dout, q, k, v, lse, delta, dk, dv, nullptr, B, Sq, Skv, Hq, Hk,              \
            (int)q_start, (float)scale, causal, stream);                                 \
      } else {                                                                           \
        launch_dkv<A, C, 4, false>(dout, q, k, v, lse, delta, dk, dv, nullptr, B, Sq,    \
                                   Skv, Hq, Hk, (int)q_start, (float)scale, causal,      \
                                   stream);                                              \
      }                                                                                  \
      if (occ_dq(A, C) == 2 && Sq % 256 == 0) {                                          \
        launch_dq<A, C, (occ_dq(A, C) == 2 ? 8 : 4), false>(                             \
            dout, q, k, v, lse, delta, dq, nullptr, B, Sq, Skv, Hq, Hk, (int)q_start,    \
            (float)scale, causal, stream);                                               \
      } else {                                                                           \
        launch_dq<A, C, 4, false>(dout, q, k, v, lse, delta, dq, nullptr, B, Sq, Skv,    \
                                  Hq, Hk, (int)q_start, (float)scale, causal, stream);   \
      }                                                                                  \
    }                                                                                    \
  }
  FA_BWD_CASE(64, 64)
  FA_BWD_CASE(96, 96)
  FA_BWD_CASE(128, 128)
  FA_BWD_CASE(192, 128)
  FA_BWD_CASE(192, 192)
  FA_BWD_CASE(256, 256)
#undef FA_BWD_CASE
  TORCH_CHECK(done, "flash_attn_bwd: unsupported head dims (Dqk=", Dqk, ", Dv=", Dv, ")");
  return {dq, dk, dv};
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

// ---- ds_read_b64_tr_b16 layout probe: fill LDS with a linear pattern and
// dump what each lane receives for reads at stride-8B lane addresses.
// Guide T10: the gather is hardware-defined; verify on silicon before
// building kernels on it (cdna_hip_programming.md §5.4 rule 27).
__global__ void tr16_probe_kernel(const bf16* __restrict__ in, float* __restrict__ out,
                                  int n) {
  __shared__ __attribute__((aligned(16))) bf16 lds[2048];
  const int l = threadIdx.x & 63;
  for (int i = threadIdx.x; i < n; i += 64) lds[i] = in[i];
  __syncthreads();
  // two reads cover 1024 B: lanes at 8-B stride (address = LDS base + offset)
  const unsigned base = (unsigned)(unsigned long long)(uintptr_t)(&lds[0]);
  for (int rd = 0; rd < 2; ++rd) {
    unsigned off = base + (unsigned)(l * 8 + rd * 512);
    unsigned long long v;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(v) : "v"(off));
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      unsigned short u = (unsigned short)(v >> (16 * j));
      out[(rd * 64 + l) * 4 + j] = bf2f(__builtin_bit_cast(bf16, u));
    }
  }
}

at::Tensor tr16_probe(const at::Tensor& pattern) {
  TORCH_CHECK(pattern.is_cuda() && pattern.scalar_type() == at::kBFloat16 &&
              pattern.numel() <= 2048, "tr16_probe: bf16, <=2048 elems");
  auto out = at::empty({2, 64, 4}, pattern.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(pattern.data_ptr()),
                     out.data_ptr<float>(), (int)pattern.numel());
  HIP_CHECK_KERNEL();
  return out;
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
