// Declarations of all HIP op entry points (implemented in the sibling .hip TUs).
#pragma once

#include <ATen/ATen.h>
#include <tuple>

namespace amd_ops {

std::tuple<at::Tensor, at::Tensor> rms_norm_fwd(const at::Tensor& x, const at::Tensor& w,
                                                double eps);
std::tuple<at::Tensor, at::Tensor> rms_norm_bwd(const at::Tensor& dy, const at::Tensor& x,
                                                const at::Tensor& w, const at::Tensor& invrms);

std::tuple<at::Tensor, at::Tensor> rope_fwd(const at::Tensor& q, const at::Tensor& k,
                                            const at::Tensor& cosb, const at::Tensor& sinb,
                                            bool backward);

at::Tensor swiglu_fwd(const at::Tensor& g, const at::Tensor& u);
at::Tensor swiglu_cat_fwd(const at::Tensor& gu);
at::Tensor swiglu_cat_bwd(const at::Tensor& dy, const at::Tensor& gu);
std::tuple<at::Tensor, at::Tensor> swiglu_bwd(const at::Tensor& dy, const at::Tensor& g,
                                              const at::Tensor& u);

void adamw_step(at::Tensor param, at::Tensor grad, at::Tensor master, at::Tensor m,
                at::Tensor v, int64_t step, double lr, double beta1, double beta2,
                double eps, double weight_decay);

std::tuple<at::Tensor, at::Tensor> flash_attn_fwd(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v, double scale,
    bool causal, int64_t q_start,
    const std::optional<at::Tensor>& doc_start = std::nullopt);
std::tuple<at::Tensor, at::Tensor, at::Tensor> flash_attn_bwd(
    const at::Tensor& dout, const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& o, const at::Tensor& lse, double scale, bool causal,
    int64_t q_start,
    const std::optional<at::Tensor>& doc_start = std::nullopt,
    const std::optional<at::Tensor>& doc_end = std::nullopt);

at::Tensor mfma_probe(const at::Tensor& a, const at::Tensor& b);
at::Tensor tr16_probe(const at::Tensor& pattern);

std::tuple<at::Tensor, at::Tensor> ce_fwd_logits(const at::Tensor& logits,
                                                 const at::Tensor& labels,
                                                 at::Tensor loss_sum);
void ce_bwd_logits(at::Tensor logits, const at::Tensor& labels, const at::Tensor& lse,
                   const at::Tensor& dloss, int64_t vocab_offset);
std::tuple<at::Tensor, at::Tensor> ce_stats_logits(const at::Tensor& logits);

std::tuple<at::Tensor, at::Tensor> fused_ce_fwd(const at::Tensor& hidden,
                                                const at::Tensor& weight,
                                                const at::Tensor& labels);

std::tuple<at::Tensor, at::Tensor, at::Tensor> soft_ce_fwd(const at::Tensor& s,
                                                           const at::Tensor& t);
at::Tensor soft_ce_bwd(const at::Tensor& s, const at::Tensor& t,
                       const at::Tensor& lse_s, const at::Tensor& lse_t,
                       const at::Tensor& dloss);

at::Tensor nf4_dequant(const at::Tensor& packed, const at::Tensor& absmax,
                       int64_t block_size, int64_t rows, int64_t cols);
at::Tensor fp8_cast(const at::Tensor& x, const at::Tensor& scale, at::Tensor amax_out,
                    bool e5m2);
at::Tensor fp8_transpose(const at::Tensor& x8);
at::Tensor lora_fused_fwd(const at::Tensor& x, const at::Tensor& A,
                          const at::Tensor& B, double scale);
at::Tensor sgmv_fused_fwd(const at::Tensor& x, const at::Tensor& A,
                          const at::Tensor& B, const at::Tensor& scales,
                          const at::Tensor& offs, const at::Tensor& tile_map,
                          const std::optional<at::Tensor>& n_tiles = std::nullopt);

at::Tensor gemv_bf16(const at::Tensor& x, const at::Tensor& w,
                     const c10::optional<at::Tensor>& bias);

std::tuple<at::Tensor, at::Tensor, at::Tensor> build_group_plan(
    const at::Tensor& counts, int64_t M, int64_t bm = 128);
at::Tensor grouped_gemm_nt(const at::Tensor& x, const at::Tensor& w,
                           const at::Tensor& offs, const at::Tensor& tile_map,
                           const std::optional<at::Tensor>& n_tiles = std::nullopt,
                           int64_t bm = 128);
at::Tensor grouped_gemm_nn(const at::Tensor& g, const at::Tensor& w,
                           const at::Tensor& offs, const at::Tensor& tile_map,
                           const std::optional<at::Tensor>& n_tiles = std::nullopt);
at::Tensor grouped_gemm_tn(const at::Tensor& g, const at::Tensor& x,
                           const at::Tensor& offs, int64_t E);
at::Tensor grouped_gemm_nt_fp8(const at::Tensor& x8, const at::Tensor& w8,
                               const at::Tensor& offs, const at::Tensor& tile_map,
                               const at::Tensor& scale,
                               const std::optional<at::Tensor>& n_tiles = std::nullopt,
                               int64_t bm = 128);
at::Tensor transpose_bf16(const at::Tensor& in);
at::Tensor permute_gather(const at::Tensor& x, const at::Tensor& src);
at::Tensor unpermute_combine(const at::Tensor& yp, const at::Tensor& pos,
                             const at::Tensor& probs);

}  // namespace amd_ops
