// TORCH_LIBRARY registration for the amd_ops namespace.
// Loaded from python via torch.ops.load_library (automodel_amd/ops/_backend.py).

#include <torch/library.h>

#include "ops_api.h"

TORCH_LIBRARY(amd_ops, m) {
  m.def("rms_norm_fwd(Tensor x, Tensor w, float eps) -> (Tensor, Tensor)");
  m.impl("rms_norm_fwd", &amd_ops::rms_norm_fwd);
  m.def("rms_norm_bwd(Tensor dy, Tensor x, Tensor w, Tensor invrms) -> (Tensor, Tensor)");
  m.impl("rms_norm_bwd", &amd_ops::rms_norm_bwd);

  m.def("rope_fwd(Tensor q, Tensor k, Tensor cos, Tensor sin, bool backward) -> (Tensor, Tensor)");
  m.impl("rope_fwd", &amd_ops::rope_fwd);

  m.def("swiglu_fwd(Tensor g, Tensor u) -> Tensor");
  m.impl("swiglu_fwd", &amd_ops::swiglu_fwd);
  m.def("swiglu_bwd(Tensor dy, Tensor g, Tensor u) -> (Tensor, Tensor)");
  m.impl("swiglu_bwd", &amd_ops::swiglu_bwd);
  m.def("swiglu_cat_fwd(Tensor gu) -> Tensor");
  m.impl("swiglu_cat_fwd", &amd_ops::swiglu_cat_fwd);
  m.def("swiglu_cat_bwd(Tensor dy, Tensor gu) -> Tensor");
  m.impl("swiglu_cat_bwd", &amd_ops::swiglu_cat_bwd);

  m.def(
      "adamw_step(Tensor(a!) param, Tensor grad, Tensor(b!) master, Tensor(c!) m, "
      "Tensor(d!) v, int step, float lr, float beta1, float beta2, float eps, "
      "float weight_decay) -> ()");
  m.impl("adamw_step", &amd_ops::adamw_step);

  m.def(
      "flash_attn_fwd(Tensor q, Tensor k, Tensor v, float scale, bool causal, "
      "int q_start=0, Tensor? doc_start=None) -> (Tensor, Tensor)");
  m.impl("flash_attn_fwd", &amd_ops::flash_attn_fwd);
  m.def(
      "flash_attn_bwd(Tensor dout, Tensor q, Tensor k, Tensor v, Tensor o, Tensor lse, "
      "float scale, bool causal, int q_start=0, Tensor? doc_start=None, "
      "Tensor? doc_end=None) -> (Tensor, Tensor, Tensor)");
  m.impl("flash_attn_bwd", &amd_ops::flash_attn_bwd);

  m.def("mfma_probe(Tensor a, Tensor b) -> Tensor");
  m.impl("mfma_probe", &amd_ops::mfma_probe);
  m.def("tr16_probe(Tensor pattern) -> Tensor");
  m.impl("tr16_probe", &amd_ops::tr16_probe);

  m.def("ce_fwd_logits(Tensor logits, Tensor labels, Tensor(a!) loss_sum) -> (Tensor, Tensor)");
  m.impl("ce_fwd_logits", &amd_ops::ce_fwd_logits);
  m.def("ce_bwd_logits(Tensor(a!) logits, Tensor labels, Tensor lse, Tensor dloss, int vocab_offset=0) -> ()");
  m.impl("ce_bwd_logits", &amd_ops::ce_bwd_logits);
  m.def("ce_stats_logits(Tensor logits) -> (Tensor, Tensor)");
  m.impl("ce_stats_logits", &amd_ops::ce_stats_logits);

  m.def("fused_ce_fwd(Tensor hidden, Tensor weight, Tensor labels) -> (Tensor, Tensor)");
  m.impl("fused_ce_fwd", &amd_ops::fused_ce_fwd);

  m.def("soft_ce_fwd(Tensor s, Tensor t) -> (Tensor, Tensor, Tensor)");
  m.impl("soft_ce_fwd", &amd_ops::soft_ce_fwd);
  m.def("soft_ce_bwd(Tensor s, Tensor t, Tensor lse_s, Tensor lse_t, Tensor dloss) -> Tensor");
  m.impl("soft_ce_bwd", &amd_ops::soft_ce_bwd);

  m.def("nf4_dequant(Tensor packed, Tensor absmax, int block_size, int rows, int cols) -> Tensor");
  m.impl("nf4_dequant", &amd_ops::nf4_dequant);
  m.def("fp8_cast(Tensor x, Tensor scale, Tensor(a!) amax_out, bool e5m2) -> Tensor");
  m.impl("fp8_cast", &amd_ops::fp8_cast);
  m.def("fp8_transpose(Tensor x8) -> Tensor");
  m.impl("fp8_transpose", &amd_ops::fp8_transpose);
  m.def("lora_fused_fwd(Tensor x, Tensor A, Tensor B, float scale) -> Tensor");
  m.impl("lora_fused_fwd", &amd_ops::lora_fused_fwd);
  m.def("sgmv_fused_fwd(Tensor x, Tensor A, Tensor B, Tensor scales, Tensor offs, Tensor tile_map, Tensor? n_tiles=None) -> Tensor");
  m.impl("sgmv_fused_fwd", &amd_ops::sgmv_fused_fwd);

  m.def("gemv_bf16(Tensor x, Tensor w, Tensor? bias) -> Tensor");
  m.impl("gemv_bf16", &amd_ops::gemv_bf16);

  m.def("build_group_plan(Tensor counts, int M, int bm=128) -> (Tensor, Tensor, Tensor)");
  m.impl("build_group_plan", &amd_ops::build_group_plan);
  m.def("grouped_gemm_nt(Tensor x, Tensor w, Tensor offs, Tensor tile_map, Tensor? n_tiles=None, int bm=128) -> Tensor");
  m.impl("grouped_gemm_nt", &amd_ops::grouped_gemm_nt);
  m.def("grouped_gemm_nn(Tensor g, Tensor w, Tensor offs, Tensor tile_map, Tensor? n_tiles=None) -> Tensor");
  m.impl("grouped_gemm_nn", &amd_ops::grouped_gemm_nn);
  m.def("grouped_gemm_tn(Tensor g, Tensor x, Tensor offs, int E) -> Tensor");
  m.impl("grouped_gemm_tn", &amd_ops::grouped_gemm_tn);
  m.def("grouped_gemm_nt_fp8(Tensor x8, Tensor w8, Tensor offs, Tensor tile_map, Tensor scale, Tensor? n_tiles=None, int bm=128) -> Tensor");
  m.impl("grouped_gemm_nt_fp8", &amd_ops::grouped_gemm_nt_fp8);
  m.def("transpose_bf16(Tensor x) -> Tensor");
  m.impl("transpose_bf16", &amd_ops::transpose_bf16);
  m.def("permute_gather(Tensor x, Tensor src) -> Tensor");
  m.impl("permute_gather", &amd_ops::permute_gather);
  m.def("unpermute_combine(Tensor yp, Tensor pos, Tensor probs) -> Tensor");
  m.impl("unpermute_combine", &amd_ops::unpermute_combine);
}
