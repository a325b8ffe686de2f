"""LoRA adapters for stacked MoE expert weights.

Reference behavior: nemo_automodel/components/_peft/lora_experts.py (per-
expert low-rank adapters on the grouped gate/up/down projections with the
base experts frozen). Adapters here are stacked like the base weights —
lora_A [E, r, in], lora_B [E, out, r] — so the adapter matmuls ride the
SAME grouped-GEMM path as the experts (two extra grouped_linear calls per
projection on GPU; einsum-free per-expert loop on CPU).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from automodel_amd.moe.experts import GroupedExperts


class GroupedExpertsLoRA(nn.Module):
    """Wraps a frozen GroupedExperts; output = base(x) + B(A(x)) * scale
    per projection. Exposes forward_permuted/forward with the base API so
    MoE modules can swap it in transparently."""

    def __init__(self, base: GroupedExperts, dim: int, alpha: float):
        super().__init__()
        self.base = base
        self.dim = dim
        self.scale = alpha / dim
        E, H, I = base.n_experts, base.hidden_size, base.intermediate_size
        self.n_experts, self.hidden_size, self.intermediate_size = E, H, I
        p = base.gate_proj
        kw = {"device": p.device, "dtype": p.dtype}
        self.lora_A_gate = nn.Parameter(torch.empty(E, dim, H, **kw))
        self.lora_B_gate = nn.Parameter(torch.zeros(E, I, dim, **kw))
        self.lora_A_up = nn.Parameter(torch.empty(E, dim, H, **kw))
        self.lora_B_up = nn.Parameter(torch.zeros(E, I, dim, **kw))
        self.lora_A_down = nn.Parameter(torch.empty(E, dim, I, **kw))
        self.lora_B_down = nn.Parameter(torch.zeros(E, H, dim, **kw))
        for a in (self.lora_A_gate, self.lora_A_up, self.lora_A_down):
            if not a.is_meta:
                nn.init.kaiming_uniform_(a, a=math.sqrt(5))
        for prm in self.base.parameters():
            prm.requires_grad_(False)

    def _adapter(self, x: torch.Tensor, A: torch.Tensor, B: torch.Tensor,
                 counts) -> torch.Tensor:
        proj = GroupedExperts.project
        return proj(proj(x, A, counts), B, counts) * self.scale

    def forward_permuted(self, x_perm: torch.Tensor, counts) -> torch.Tensor:
        from automodel_amd.ops.swiglu import swiglu

        base = self.base
        proj = GroupedExperts.project
        if x_perm.numel() == 0:
            return x_perm[:0]
        g = proj(x_perm, base.gate_proj, counts) + self._adapter(
            x_perm, self.lora_A_gate, self.lora_B_gate, counts)
        u = proj(x_perm, base.up_proj, counts) + self._adapter(
            x_perm, self.lora_A_up, self.lora_B_up, counts)
        h = swiglu(g, u)
        return proj(h, base.down_proj, counts) + self._adapter(
            h, self.lora_A_down, self.lora_B_down, counts)

    def forward(self, x: torch.Tensor, probs: torch.Tensor,
                indices: torch.Tensor) -> torch.Tensor:
        from automodel_amd.moe.experts import permute_tokens, unpermute_tokens

        x_perm, sort_idx, counts = permute_tokens(x, indices, self.n_experts)
        y_perm = self.forward_permuted(x_perm, counts)
        return unpermute_tokens(y_perm, sort_idx, probs)


def apply_lora_to_grouped_experts(model: nn.Module, dim: int = 8,
                                  alpha: float = 16.0) -> int:
    """Swap every GroupedExperts for GroupedExpertsLoRA. Returns count."""
    n = 0
    for name, parent in list(model.named_modules()):
        for child_name, child in list(parent.named_children()):
            if type(child) is GroupedExperts:
                setattr(parent, child_name, GroupedExpertsLoRA(child, dim, alpha))
                n += 1
    return n
