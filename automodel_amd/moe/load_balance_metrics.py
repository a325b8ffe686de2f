"""MoE expert load-balance metrics (reference moe/load_balance_metrics.py:396).

Collects per-layer expert loads from MoE modules, all-reduces over the DP
group, and produces brief (global imbalance) or detailed (per-layer) stats
for the metric logger.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from automodel_amd.moe.layers import MoE


def collect_expert_loads(model) -> torch.Tensor | None:
    """Stack per-layer loads [L, E] from the last forward (None if no MoE)."""
    loads = [m.last_expert_load for m in model.modules()
             if isinstance(m, MoE) and m.last_expert_load is not None]
    if not loads:
        return None
    return torch.stack(loads)


def load_balance_metrics(model, group=None, detailed: bool = False) -> dict:
    loads = collect_expert_loads(model)
    if loads is None:
        return {}
    if dist.is_available() and dist.is_initialized() and dist.get_world_size(group) > 1:
        dist.all_reduce(loads, group=group)
    frac = loads / loads.sum(dim=-1, keepdim=True).clamp_min(1)
    E = loads.shape[-1]
    # max-violation ratio: max load / ideal uniform load
    imbalance = (frac.max(dim=-1).values * E)
    out = {
        "moe_imbalance_mean": float(imbalance.mean()),
        "moe_imbalance_max": float(imbalance.max()),
    }
    if detailed:
        for i in range(loads.shape[0]):
            out[f"moe_layer{i}_max_frac"] = float(frac[i].max())
    return out
