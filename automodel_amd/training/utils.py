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


def _grad_sq_contribution(g: torch.Tensor, world: int) -> torch.Tensor:
    """This rank's contribution to the global sq-norm such that a WORLD
    all-reduce counts every element exactly once (reference
    training/utils.py:400: replication-aware grad-norm)."""
    if isinstance(g, DTensor):
        local = g.to_local()
        shard_ranks = 1
        for dim_size, placement in zip(g.device_mesh.shape, g.placements):
            if placement.is_shard():
                shard_ranks *= dim_size
        replicas = max(1, world // shard_ranks)
        return local.float().pow(2).sum() / replicas
    # plain tensors are replicated on every rank
    sq = g.float().pow(2).sum()
    return sq / world if world > 1 else sq


def clip_grad_norm_(parameters, max_norm: float, group=None) -> torch.Tensor:
    """Global 2-norm clip correct for DTensor grads across mixed shardings:
    sharded elements are counted once via their owning shard; replicated
    elements are divided by their replica count before the all-reduce."""
    params = [p for p in parameters if p.grad is not None]
    if not params:
        return torch.tensor(0.0)
    device = params[0].grad.device
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    total_sq = torch.zeros((), dtype=torch.float32, device=device)
    for p in params:
        total_sq += _grad_sq_contribution(p.grad, world)
    if dist.is_initialized() and world > 1:
        dist.all_reduce(total_sq, group=group)
    total_norm = total_sq.sqrt()
    if max_norm is not None and max_norm > 0:
        clip_coef = torch.clamp(max_norm / (total_norm + 1e-6), max=1.0)
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
