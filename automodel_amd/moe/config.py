"""MoE configuration (reference: nemo_automodel/components/moe/config.py:97)."""

from __future__ import annotations

from dataclasses import dataclass


@dataclass
class MoEConfig:
    n_routed_experts: int = 8
    n_shared_experts: int = 0
    n_activated_experts: int = 2          # top-k
    n_expert_groups: int = 1
    n_limited_groups: int = 1
    score_func: str = "softmax"           # softmax | sigmoid
    route_scale: float = 1.0
    aux_loss_coeff: float = 0.0
    norm_topk_prob: bool = True
    expert_bias: bool = False             # aux-free balancing bias (DeepSeek)
    bias_update_speed: float = 1e-3
    moe_intermediate_size: int | None = None
    shared_expert_intermediate_size: int | None = None
    shared_expert_gate: bool = False   # qwen2-moe: sigmoid-gated shared expert
    topk_then_softmax: bool = False    # granite-moe: topk raw logits, softmax over k
    fake_balanced_gate: bool = False      # benchmark ideal routing
