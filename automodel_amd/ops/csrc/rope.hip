// Fused rotary embedding for q and k (CDNA4).
// HF rotate-half convention: y1 = x1*cos - x2*sin ; y2 = x2*cos + x1*sin over
// the two halves of head_dim. cos/sin tables are host-precomputed fp32
// (guide Appendix B: no device trig). One launch covers q AND k.
// Replaces the reference's TE/Liger fused-RoPE backends.

#include <torch/library.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

// Each thread handles TWO adjacent pairs (4B loads from each half-row).
// rows = B*S*Hq followed by B*S*Hk; table row = s = (flat_row / H) % S.
__global__ void rope_kernel(const bf16* __restrict__ q, const bf16* __restrict__ k,
                            const float* __restrict__ cosb, const float* __restrict__ sinb,
                            bf16* __restrict__ qo, bf16* __restrict__ ko,
                            int S, int Hq, int Hk, int D, long q_rows, long total_rows,
                            float sign) {
  const int half = D / 2;
  const int pairs2 = half / 2;  // 2-pair work items per row
  const long gid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long row = gid / pairs2;
  if (row >= total_rows) return;
  const int p2 = (int)(gid % pairs2);
  const int j = p2 * 2;

  const bf16* src;
  bf16* dst;
  long r;
  int H;
  if (row < q_rows) {
    r = row; H = Hq; src = q; dst = qo;
  } else {
    r = row - q_rows; H = Hk; src = k; dst = ko;
  }
  const int s = (int)((r / H) % S);
  const long base = r * (long)D;

  // 4-byte vector loads: two bf16 from each half
  ushort2 x1u = *reinterpret_cast<const ushort2*>(src + base + j);
  ushort2 x2u = *reinterpret_cast<const ushort2*>(src + base + j + half);
  float2 c = *reinterpret_cast<const float2*>(cosb + (long)s * D + j);
  float2 sn = *reinterpret_cast<const float2*>(sinb + (long)s * D + j);
  sn.x *= sign; sn.y *= sign;

  bf16 x1a = *reinterpret_cast<bf16*>(&x1u.x), x1b = *reinterpret_cast<bf16*>(&x1u.y);
  bf16 x2a = *reinterpret_cast<bf16*>(&x2u.x), x2b = *reinterpret_cast<bf16*>(&x2u.y);
  float f1a = bf2f(x1a), f1b = bf2f(x1b), f2a = bf2f(x2a), f2b = bf2f(x2b);

  bf16 y1a = f2bf(f1a * c.x - f2a * sn.x);
  bf16 y1b = f2bf(f1b * c.y - f2b * sn.y);
  bf16 y2a = f2bf(f2a * c.x + f1a * sn.x);
  bf16 y2b = f2bf(f2b * c.y + f1b * sn.y);

  ushort2 y1u{*reinterpret_cast<unsigned short*>(&y1a), *reinterpret_cast<unsigned short*>(&y1b)};
  ushort2 y2u{*reinterpret_cast<unsigned short*>(&y2a), *reinterpret_cast<unsigned short*>(&y2b)};
  *reinterpret_cast<ushort2*>(dst + base + j) = y1u;
  *reinterpret_cast<ushort2*>(dst + base + j + half) = y2u;
}

// q: [B,S,Hq,D] bf16, k: [B,S,Hk,D] bf16, cos/sin: [S,D] f32.
std::tuple<at::Tensor, at::Tensor> rope_fwd(const at::Tensor& q, const at::Tensor& k,
                                            const at::Tensor& cosb, const at::Tensor& sinb,
                                            bool backward) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.scalar_type() == at::kBFloat16,
              "rope_fwd: q must be 4-D bf16");
  TORCH_CHECK(cosb.scalar_type() == at::kFloat && cosb.dim() == 2, "rope: cos must be [S,D] f32");
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hk = k.size(2);
  TORCH_CHECK(cosb.size(0) >= S && cosb.size(1) == D, "rope: table shape mismatch");
  TORCH_CHECK(D % 4 == 0, "rope: head_dim must be a multiple of 4");
  auto qo = at::empty_like(q);
  auto ko = at::empty_like(k);
  const long q_rows = (long)B * S * Hq;
  const long total_rows = q_rows + (long)B * S * Hk;
  const long work = total_rows * (D / 4);
  const int block = 256;
  const long grid = (work + block - 1) / block;
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rope_kernel, dim3((unsigned)grid), dim3(block), 0, stream.stream(),
                     reinterpret_cast<const bf16*>(q.data_ptr()),
                     reinterpret_cast<const bf16*>(k.data_ptr()),
                     cosb.data_ptr<float>(), sinb.data_ptr<float>(),
                     reinterpret_cast<bf16*>(qo.data_ptr()),
                     reinterpret_cast<bf16*>(ko.data_ptr()),
                     S, Hq, Hk, D, q_rows, total_rows, backward ? -1.f : 1.f);
  HIP_CHECK_KERNEL();
  return {qo, ko};
}

}  // namespace amd_ops
