"""Decode-path linear: route skinny (rows<=16) matmuls to the HIP GEMV.

Reference behavior: serving engines (which the reference defers to for
deployment) special-case decode GEMVs; here the kernel is in-tree
(ops/csrc/gemv.hip — wave-per-row bf16 row streaming) and DecodeLinear
swaps in for inference. Training paths are untouched: the GEMV route only
fires under no_grad with a 3-D [B, 1, K] or 2-D [<=16, K] bf16 input on
GPU; everything else falls through to F.linear (hipBLASLt)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

_PAD_TO = {5: 8, 6: 8, 7: 8, 9: 16, 10: 16, 11: 16, 12: 16, 13: 16, 14: 16, 15: 16}


def gemv_bf16(x2d: torch.Tensor, w: torch.Tensor,
              bias: torch.Tensor | None = None) -> torch.Tensor:
    """y = x2d @ w.T via the HIP kernel; pads odd batch sizes to a
    compiled template instance."""
    from automodel_amd.ops._backend import hip_ops

    B = x2d.shape[0]
    tgt = _PAD_TO.get(B, B)
    if tgt != B:
        x2d = F.pad(x2d, (0, 0, 0, tgt - B))
    y = hip_ops().gemv_bf16(x2d, w, bias)
    return y[:B] if tgt != B else y


def _gemv_eligible(x: torch.Tensor, w: torch.Tensor) -> bool:
    if torch.is_grad_enabled() or not x.is_cuda:
        return False
    if x.dtype != torch.bfloat16 or w.dtype != torch.bfloat16:
        return False
    rows = x.numel() // x.shape[-1]
    # measured on MI355X (profiles/README.md): the wave-per-row GEMV beats
    # hipBLASLt up to ~4 rows; above that the library's m-tiled path wins
    return rows <= 4 and x.shape[-1] % 512 == 0


class DecodeLinear(nn.Linear):
    """nn.Linear that runs decode-shaped inputs on the in-tree GEMV."""

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "DecodeLinear":
        m = cls(lin.in_features, lin.out_features, bias=lin.bias is not None,
                device=lin.weight.device, dtype=lin.weight.dtype)
        with torch.no_grad():
            m.weight.copy_(lin.weight)
            if lin.bias is not None:
                m.bias.copy_(lin.bias)
        return m

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _gemv_eligible(x, self.weight):
            lead = x.shape[:-1]
            y = gemv_bf16(x.reshape(-1, x.shape[-1]), self.weight, self.bias)
            return y.reshape(*lead, self.out_features)
        return F.linear(x, self.weight, self.bias)


def swap_linears_for_decode(model: nn.Module) -> int:
    """Swap every nn.Linear for DecodeLinear (weights shared by copy).
    Returns count. Call on an inference model before generate_graphed."""
    n = 0
    for name, parent in list(model.named_modules()):
        for child_name, child in list(parent.named_children()):
            if type(child) is nn.Linear:
                setattr(parent, child_name, DecodeLinear.from_linear(child))
                n += 1
    return n
