"""InfoNCE contrastive loss with cross-rank in-batch negatives.

Reference behavior: nemo_automodel/components/loss/infonce.py (retrieval
bi-encoder training; negatives gathered across the DP group).
"""

from __future__ import annotations

import torch
import torch.distributed as dist


class _GatherCat(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = dist.get_world_size(group)
        ctx.rank = dist.get_rank(group)
        ctx.local = x.shape[0]
        outs = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(outs, x.contiguous(), group=group)
        outs[ctx.rank] = x  # keep local autograd path
        return torch.cat(outs, dim=0)

    @staticmethod
    def backward(ctx, grad):
        return grad.narrow(0, ctx.rank * ctx.local, ctx.local), None


def info_nce_loss(
    query: torch.Tensor,
    positive: torch.Tensor,
    temperature: float = 0.05,
    group=None,
) -> torch.Tensor:
    """query/positive [B, D] normalized embeddings; negatives are every other
    positive in the (cross-rank) batch. Returns mean CE loss."""
    q = torch.nn.functional.normalize(query, dim=-1)
    p = torch.nn.functional.normalize(positive, dim=-1)
    offset = 0
    if group is not None and dist.is_initialized() and dist.get_world_size(group) > 1:
        offset = dist.get_rank(group) * q.shape[0]
        p = _GatherCat.apply(p, group)
    logits = (q @ p.t()) / temperature
    labels = torch.arange(q.shape[0], device=q.device) + offset
    return torch.nn.functional.cross_entropy(logits, labels)
