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
            if dispatcher == "a2a_pipelined":
                from automodel_amd.moe.dispatch import PipelinedAllToAllDispatcher

                m.dispatcher = PipelinedAllToAllDispatcher(group, E, offset, n_local)
            else:
                disp_cls = (AllToAllDispatcher if dispatcher == "a2a"
                            else AllGatherDispatcher)
                m.dispatcher = disp_cls(group, E, offset, n_local)
    return model


def parallelize_moe_model(
    model: nn.Module,
    ep: int,
    dispatcher: str = "a2a",
    device_type: str | None = None,
    param_dtype=None,
    reshard_after_forward: bool = False,
) -> nn.Module:
    """Compose EP with FSDP (reference moe/parallelizer.py:978 parallelize_model):

      1. carve the MoE mesh (ep_shard, ep) from the world,
      2. slice experts over the ep axis + attach the RCCL a2a dispatcher,
      3. fully_shard expert modules over ep_shard (each expert shard is
         further sharded only across ranks that hold the SAME experts),
      4. fully_shard the decoder layers + root over the full dp world
         (expert params are already managed by their inner FSDP module).
    """
    import torch.distributed as dist
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.fsdp import MixedPrecisionPolicy, fully_shard

    from automodel_amd.parallel.fsdp import apply_fsdp

    world = dist.get_world_size()
    assert world % ep == 0, f"world {world} not divisible by ep {ep}"
    ep_shard = world // ep
    if device_type is None:
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
    moe_mesh = init_device_mesh(device_type, (ep_shard, ep),
                                mesh_dim_names=("ep_shard", "ep"))
    apply_ep(model, moe_mesh["ep"], dispatcher=dispatcher)

    param_dtype = param_dtype or (torch.bfloat16 if device_type == "cuda" else torch.float32)
    mp = MixedPrecisionPolicy(param_dtype=param_dtype, reduce_dtype=torch.float32)
    if ep_shard > 1:
        for m in model.modules():
            if isinstance(m, MoE):
                fully_shard(m.experts, mesh=moe_mesh["ep_shard"], mp_policy=mp,
                            reshard_after_forward=reshard_after_forward)
    dp_mesh = init_device_mesh(device_type, (world,), mesh_dim_names=("dp",))
    apply_fsdp(model, dp_mesh["dp"], param_dtype=param_dtype,
               reshard_after_forward=reshard_after_forward)
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
