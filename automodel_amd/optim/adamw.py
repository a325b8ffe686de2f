"""Fused AdamW with fp32 master weights for bf16 training.

MI355X-native replacement for the reference's Apex FusedAdam / torch fused
adam path (nemo_automodel/components/optim/optimizer.py:208-320). The update
math matches torch.optim.AdamW exactly (parity-tested); bf16 CUDA params run
the hand-written HIP kernel (csrc/adamw.hip) which keeps an fp32 master copy
and writes the bf16 param in the same pass.
"""

from __future__ import annotations

import math
from typing import Iterable

import torch
from torch.distributed.tensor import DTensor

from automodel_amd.ops._backend import hip_ops


def _local(t: torch.Tensor) -> torch.Tensor:
    """Local shard view for DTensor (FSDP2) params — same storage."""
    return t.to_local() if isinstance(t, DTensor) else t


class FusedAdamW(torch.optim.Optimizer):
    def __init__(
        self,
        params: Iterable,
        lr: float = 1e-4,
        betas: tuple[float, float] = (0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
        state_dtype: torch.dtype = torch.float32,
    ):
        # state_dtype=torch.bfloat16 halves optimizer memory (m/v bf16, no
        # fp32 master) for single-GPU benches of 30B-class models; multi-GPU
        # training keeps the default fp32 states (sharded by FSDP2)
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        self.state_dtype = state_dtype
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps, wd = group["eps"], group["weight_decay"]
            for p_ in group["params"]:
                if p_.grad is None:
                    continue
                p = _local(p_)
                grad = _local(p_.grad)
                state = self.state[p_]
                use_hip = (p.is_cuda and p.dtype == torch.bfloat16
                           and p.numel() % 4 == 0)
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=self.state_dtype)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=self.state_dtype)
                    if use_hip and self.state_dtype == torch.float32:
                        state["master"] = p.detach().float().clone()
                state["step"] += 1
                t = state["step"]
                m, v = state["exp_avg"], state["exp_avg_sq"]
                if use_hip:
                    # fp32 states ride the master-weight kernel; bf16 states
                    # ride the master-free kernel (m passed as placeholder
                    # for the unused master arg)
                    hip_ops().adamw_step(
                        p, grad.to(torch.bfloat16), state.get("master", m), m, v,
                        t, lr, beta1, beta2, eps, wd,
                    )
                else:
                    # reference math (identical to torch.optim.AdamW)
                    gf = grad.float()
                    pf = p.float()
                    pf.mul_(1 - lr * wd)
                    m.mul_(beta1).add_(gf, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
                    bc1 = 1 - beta1**t
                    bc2 = 1 - beta2**t
                    denom = (v.sqrt() / math.sqrt(bc2)).add_(eps)
                    pf.addcdiv_(m, denom, value=-lr / bc1)
                    p.copy_(pf.to(p.dtype))
        return loss

    def state_dict(self):
        sd = super().state_dict()
        return sd

    def zero_grad(self, set_to_none: bool = True):
        super().zero_grad(set_to_none=set_to_none)


def build_adamw(model: torch.nn.Module | Iterable, lr: float = 1e-4,
                betas: tuple[float, float] = (0.9, 0.999), eps: float = 1e-8,
                weight_decay: float = 0.01, fused: bool = True) -> torch.optim.Optimizer:
    """Config-facing factory (``optimizer._target_: automodel_amd.optim.build_adamw``)."""
    params = model.parameters() if isinstance(model, torch.nn.Module) else model
    params = [p for p in params if p.requires_grad]
    if fused:
        return FusedAdamW(params, lr=lr, betas=tuple(betas), eps=eps, weight_decay=weight_decay)
    return torch.optim.AdamW(params, lr=lr, betas=tuple(betas), eps=eps, weight_decay=weight_decay)
