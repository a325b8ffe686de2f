"""Additional distillation / ranking losses.

Reference behavior: nemo_automodel/components/loss/{embedding_distill.py,
intermediate_distill.py, listmle.py} — embedding-space distillation (cosine /
MSE between student and teacher hidden states, with a learned projection
when widths differ), per-layer intermediate-state distillation with layer
mapping, and ListMLE listwise ranking loss for retrieval re-rankers.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class EmbeddingDistillLoss(nn.Module):
    """Distill teacher hidden states into the student's (mode: "mse" |
    "cosine"). A bias-free projection maps student width -> teacher width
    when they differ (trained jointly)."""

    def __init__(self, student_dim: int, teacher_dim: int, mode: str = "cosine"):
        super().__init__()
        assert mode in ("mse", "cosine")
        self.mode = mode
        self.proj = (nn.Linear(student_dim, teacher_dim, bias=False)
                     if student_dim != teacher_dim else nn.Identity())

    def forward(self, student_h: torch.Tensor, teacher_h: torch.Tensor,
                mask: torch.Tensor | None = None) -> torch.Tensor:
        s = self.proj(student_h).float()
        t = teacher_h.detach().float()
        if self.mode == "mse":
            per_tok = (s - t).pow(2).mean(-1)
        else:
            per_tok = 1.0 - F.cosine_similarity(s, t, dim=-1)
        if mask is not None:
            per_tok = per_tok * mask
            return per_tok.sum() / mask.sum().clamp_min(1)
        return per_tok.mean()


class IntermediateDistillLoss(nn.Module):
    """Per-layer hidden-state distillation with an explicit layer map
    {student_layer: teacher_layer}; sums EmbeddingDistillLoss over pairs."""

    def __init__(self, student_dim: int, teacher_dim: int,
                 layer_map: dict[int, int], mode: str = "mse"):
        super().__init__()
        self.layer_map = dict(layer_map)
        self.inner = EmbeddingDistillLoss(student_dim, teacher_dim, mode)

    def forward(self, student_layers: list[torch.Tensor],
                teacher_layers: list[torch.Tensor],
                mask: torch.Tensor | None = None) -> torch.Tensor:
        total = 0.0
        for s_i, t_i in self.layer_map.items():
            total = total + self.inner(student_layers[s_i], teacher_layers[t_i], mask)
        return total / max(len(self.layer_map), 1)


def listmle_loss(scores: torch.Tensor, relevance: torch.Tensor) -> torch.Tensor:
    """ListMLE listwise ranking loss: -log P(permutation sorted by relevance
    | scores) under the Plackett-Luce model. scores/relevance [B, L]; larger
    relevance = should rank earlier."""
    order = relevance.argsort(dim=-1, descending=True)
    s = scores.gather(-1, order).float()
    # log-cumsum-exp over the remaining suffix at each rank position
    rev = s.flip(-1)
    denom = torch.logcumsumexp(rev, dim=-1).flip(-1)
    return (denom - s).sum(-1).mean()
