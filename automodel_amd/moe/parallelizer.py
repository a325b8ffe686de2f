"""Expert parallelism application: shard stacked experts over the ep mesh and
attach a token dispatcher.

Reference behavior: nemo_automodel/components/moe/parallelizer.py:278-978
(ExpertParallel shards expert dim-0 over the ep mesh; parallelize_model
composes TP -> EP -> FSDP). Here experts are sliced to local shards and the
dispatcher exchanges tokens with RCCL a2a (dispatch.py); FSDP composition
wraps non-expert modules over dp and expert modules over ep_shard.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from automodel_amd.moe.dispatch import AllGatherDispatcher, AllToAllDispatcher
from automodel_amd.moe.layers import MoE


def apply_ep(model: nn.Module, ep_mesh, dispatcher: str = "a2a") -> nn.Module:
    """Slice each MoE module's experts to this rank's shard and attach the
    EP dispatcher. ep_mesh: a DeviceMesh axis (or ProcessGroup)."""
    group = ep_mesh.get_group() if hasattr(ep_mesh, "get_group") else ep_mesh
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if world == 1:
        return model
    for m in model.modules():
        if isinstance(m, MoE):
            E = m.experts.n_experts
            assert E % world == 0, f"{E} experts not divisible by ep={world}"
            n_local = E // world
            offset = rank * n_local
            with torch.no_grad():
                for name in ("gate_proj", "up_proj", "down_proj"):
                    full = getattr(m.experts, name)
                    local = nn.Parameter(full[offset : offset + n_local].clone())
                    setattr(m.experts, name, local)
            m.experts.n_experts = n_local
            disp_cls = AllToAllDispatcher if dispatcher == "a2a" else AllGatherDispatcher
            m.dispatcher = disp_cls(group, E, offset, n_local)
    return model


def moe_param_groups(model: nn.Module) -> tuple[list, list]:
    """(expert_params, dense_params) — expert grads are NOT reduced over dp
    when EP spans the dp group (each rank owns distinct experts)."""
    expert_params, dense = [], []
    expert_modules = [m.experts for m in model.modules() if isinstance(m, MoE)]
    expert_ids = {id(p) for em in expert_modules for p in em.parameters()}
    for p in model.parameters():
        (expert_params if id(p) in expert_ids else dense).append(p)
    return expert_params, dense
