"""RMSNorm with a CDNA4 HIP kernel (fwd+bwd) and a torch reference path.

Replaces the reference's TE/QuACK/Liger RMSNorm backends
(nemo_automodel/components/models/common/utils.py:282, kernel_patches.py:155)
with one hand-written HIP kernel (csrc/rms_norm.hip).
"""

from __future__ import annotations

import torch

from automodel_amd.ops._backend import hip_ops


def rms_norm_ref(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """fp32-upcast reference (used on CPU and in kernel parity tests)."""
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv).to(x.dtype) * weight


class _RMSNormHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, eps: float):
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        y, invrms = hip_ops().rms_norm_fwd(x2d, weight, eps)
        ctx.save_for_backward(x2d, weight, invrms)
        ctx.x_shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x2d, weight, invrms = ctx.saved_tensors
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx, dw = hip_ops().rms_norm_bwd(dy2d, x2d, weight, invrms)
        return dx.view(ctx.x_shape), dw.to(weight.dtype), None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6, backend: str = "hip") -> torch.Tensor:
    if backend == "hip" and x.is_cuda:
        return _RMSNormHip.apply(x, weight, eps)
    return rms_norm_ref(x, weight, eps)


class RMSNorm(torch.nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-6, backend: str = "hip",
                 device=None, dtype=None):
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(hidden_size, device=device, dtype=dtype))
        self.variance_epsilon = eps
        self.backend = backend

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return rms_norm(x, self.weight, self.variance_epsilon, self.backend)

    def reset_parameters(self) -> None:
        torch.nn.init.ones_(self.weight)

    def extra_repr(self) -> str:
        return f"{self.weight.shape[0]}, eps={self.variance_epsilon}, backend={self.backend}"
