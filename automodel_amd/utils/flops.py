"""Analytic FLOPs + MFU calculation.

Reference behavior: nemo_automodel/components/utils/flops_utils.py:19-650
(per-family analytic FLOPs; MFU = achieved / peak). MI355X denominator:
peak dense bf16 ~= 2.5 PFLOP/s per GPU (CDNA4; AMD's larger headline figures
include 2:1 structured sparsity — guide §5.4 rule 11).
"""

from __future__ import annotations

MI355X_PEAK_BF16 = 2.5e15  # dense
MI355X_PEAK_FP8 = 5.0e15


def llama_flops_per_token(
    hidden: int,
    intermediate: int,
    layers: int,
    vocab: int,
    seq_len: int,
    num_heads: int,
    num_kv_heads: int,
    head_dim: int | None = None,
) -> float:
    """Training FLOPs per token (fwd+bwd = 3x fwd) for a Llama-style model.

    Matches the reference's llama3 calculator structure (flops_utils.py:95):
    attention projections + scores + MLP + LM head; causal attention counted
    at S/2 average context.
    """
    d = head_dim or hidden // num_heads
    q_size = num_heads * d
    kv_size = num_kv_heads * d
    # per-layer, per-token MACs
    attn_proj = hidden * q_size + 2 * hidden * kv_size + q_size * hidden
    attn_scores = 2 * (seq_len / 2) * d * num_heads  # QK^T + PV at avg causal ctx
    mlp = 3 * hidden * intermediate
    per_layer = attn_proj + attn_scores + mlp
    lm_head = hidden * vocab
    fwd = 2 * (layers * per_layer + lm_head)  # 2 FLOPs per MAC
    return 3.0 * fwd  # fwd + bwd (2x)


def mfu(tokens_per_sec_per_gpu: float, flops_per_token: float,
        peak: float = MI355X_PEAK_BF16) -> float:
    return tokens_per_sec_per_gpu * flops_per_token / peak


def moe_flops_per_token(
    hidden: int, layers: int, vocab: int, seq_len: int,
    num_heads: int, num_kv_heads: int,
    moe_intermediate: int, n_activated: int, n_shared: int = 0,
    shared_intermediate: int | None = None, head_dim: int | None = None,
) -> float:
    """Training FLOPs/token for MoE models: only ACTIVATED experts count
    (reference flops_utils.py mixtral/deepseekv3 calculators)."""
    d = head_dim or hidden // num_heads
    attn_proj = hidden * num_heads * d * 2 + 2 * hidden * num_kv_heads * d
    attn_scores = 2 * (seq_len / 2) * d * num_heads
    expert_mlp = 3 * hidden * moe_intermediate * n_activated
    shared_mlp = 3 * hidden * (shared_intermediate or moe_intermediate) * (1 if n_shared else 0)
    per_layer = attn_proj + attn_scores + expert_mlp + shared_mlp
    fwd = 2 * (layers * per_layer + hidden * vocab)
    return 3.0 * fwd


def deepseek_v3_flops_per_token(cfg, seq_len: int) -> float:
    """Training FLOPs/token for MLA + MoE models (reference flops_utils.py:437).
    cfg: DeepseekV3Config."""
    H = cfg.hidden_size
    nh = cfg.num_attention_heads
    qk, vd = cfg.qk_head_dim, cfg.v_head_dim
    # MLA projections per token
    q_proj = (H * cfg.q_lora_rank + cfg.q_lora_rank * nh * qk) if cfg.q_lora_rank \
        else H * nh * qk
    kv_proj = H * (cfg.kv_lora_rank + cfg.qk_rope_head_dim) + \
        cfg.kv_lora_rank * nh * (cfg.qk_nope_head_dim + vd)
    o_proj = nh * vd * H
    scores = 2 * (seq_len / 2) * nh * (qk + vd) / 2  # QK^T + PV at avg ctx
    moe = cfg.moe
    dense_layers = cfg.first_k_dense_replace
    moe_layers = cfg.num_hidden_layers - dense_layers
    dense_mlp = 3 * H * cfg.intermediate_size
    inter = moe.moe_intermediate_size or H
    moe_mlp = 3 * H * inter * moe.n_activated_experts + \
        3 * H * (moe.shared_expert_intermediate_size or inter) * (1 if moe.n_shared_experts else 0)
    attn = q_proj + kv_proj + o_proj + scores
    total = (dense_layers * (attn + dense_mlp) + moe_layers * (attn + moe_mlp)
             + H * cfg.vocab_size)
    return 3.0 * 2 * total
