"""Training utilities: grad clipping across meshes, grad-accum helpers.

Reference behavior: nemo_automodel/components/training/utils.py:329-450
(prepare_for_grad_accumulation defers FSDP grad sync on non-final
microbatches; scale_grads_and_clip_grad_norm computes a global 2-norm with
sharding-aware reductions).
"""

from __future__ import annotations

import torch
import torch.distributed as dist
from torch.distributed.tensor import DTensor


def clip_grad_norm_(parameters, max_norm: float, group=None) -> torch.Tensor:
    """Global 2-norm clip correct for DTensor (FSDP2-sharded) grads.

    Local shard sq-sums are all-reduced over ``group`` (default: WORLD, which
    is correct for pure FSDP sharding where every rank holds a disjoint shard).
    """
    params = [p for p in parameters if p.grad is not None]
    if not params:
        return torch.tensor(0.0)
    device = params[0].grad.device
    total_sq = torch.zeros((), dtype=torch.float32, device=device)
    for p in params:
        g = p.grad
        local = g.to_local() if isinstance(g, DTensor) else g
        total_sq += local.float().pow(2).sum()
    if dist.is_initialized() and dist.get_world_size(group) > 1:
        dist.all_reduce(total_sq, group=group)
    total_norm = total_sq.sqrt()
    if max_norm is not None and max_norm > 0:
        clip_coef = max_norm / (total_norm + 1e-6)
        clip_coef = torch.clamp(clip_coef, max=1.0)
        for p in params:
            g = p.grad
            local = g.to_local() if isinstance(g, DTensor) else g
            local.mul_(clip_coef.to(local.dtype))
    return total_norm


def prepare_for_grad_accumulation(model: torch.nn.Module, is_final_microbatch: bool) -> None:
    """Defer FSDP2 reduce-scatter until the final microbatch
    (reference training/utils.py:329)."""
    if hasattr(model, "set_requires_gradient_sync"):
        model.set_requires_gradient_sync(is_final_microbatch)


def count_label_tokens(labels: torch.Tensor, ignore_index: int = -100) -> torch.Tensor:
    return (labels != ignore_index).sum()
