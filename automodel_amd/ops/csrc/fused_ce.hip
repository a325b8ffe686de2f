// Fused linear cross-entropy (lm_head GEMM + online softmax, no logits tensor).
// MI355X-native cut-cross-entropy equivalent (reference: loss/linear_ce.py:253).
//
// STATUS: HIP kernel in progress — the python wrapper currently routes GPU
// calls through the chunked hipBLASLt path (loss/linear_ce.py) which already
// avoids materializing the full [T, V] logits tensor. These entry points fail
// loudly so nothing silently falls back through them.

#include <torch/library.h>
#include <ATen/ATen.h>

#include "common.h"
#include "ops_api.h"

namespace amd_ops {

std::tuple<at::Tensor, at::Tensor> fused_ce_fwd(const at::Tensor& hidden,
                                                const at::Tensor& weight,
                                                const at::Tensor& labels) {
  TORCH_CHECK(false, "fused_ce_fwd HIP kernel not built yet — use loss backend 'chunked'");
}

std::tuple<at::Tensor, at::Tensor> fused_ce_bwd(const at::Tensor& hidden,
                                                const at::Tensor& weight,
                                                const at::Tensor& labels,
                                                const at::Tensor& lse,
                                                const at::Tensor& dloss) {
  TORCH_CHECK(false, "fused_ce_bwd HIP kernel not built yet — use loss backend 'chunked'");
}

}  // namespace amd_ops
