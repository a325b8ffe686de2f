"""FP8 training: dynamic tensorwise-scaled Float8Linear on MFMA.

Reference behavior: nemo_automodel/components/quantization/fp8.py:130
(torchao float8 linear swap with tensorwise/rowwise recipes). MI355X-native
implementation over torch._scaled_mm (hipBLASLt fp8 MFMA, ~2x the bf16 rate;
gfx950 uses OCP e4m3fn/e5m2 — guide §4):

  forward:  y  = x_e4m3 @ W_e4m3^T
  dgrad:    dx = g_e5m2 @ W_e4m3
  wgrad:    dW = g_e5m2^T @ x_e4m3

Scales are per-tensor dynamic (amax / dtype_max). Linears with dims not
divisible by 16 (or tiny layers) are left in bf16.
"""

from __future__ import annotations

import torch
import torch.nn as nn

E4M3_MAX = 448.0
E5M2_MAX = 57344.0


def _cast_fp8(t: torch.Tensor, dtype, max_val: float):
    amax = t.abs().amax().clamp(min=1e-12).float()
    scale = (max_val / amax).clamp(max=1e12)
    t8 = (t.float() * scale).clamp(-max_val, max_val).to(dtype)
    return t8, scale.reciprocal()  # returns inverse scale (dequant factor)


def _scaled_mm(a8, b8, inv_a, inv_b, out_dtype=torch.bfloat16):
    return torch._scaled_mm(a8, b8, scale_a=inv_a, scale_b=inv_b,
                            out_dtype=out_dtype)


class _Fp8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        x8, inv_x = _cast_fp8(x2, torch.float8_e4m3fn, E4M3_MAX)
        w8, inv_w = _cast_fp8(weight, torch.float8_e4m3fn, E4M3_MAX)
        y = _scaled_mm(x8, w8.t(), inv_x, inv_w, out_dtype=x.dtype)
        if bias is not None:
            y = y + bias
        ctx.save_for_backward(x8, inv_x, w8, inv_w)
        ctx.x_shape = shape
        ctx.has_bias = bias is not None
        return y.view(*shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, g):
        x8, inv_x, w8, inv_w = ctx.saved_tensors
        g2 = g.reshape(-1, g.shape[-1])
        g8, inv_g = _cast_fp8(g2, torch.float8_e5m2, E5M2_MAX)
        # dx = g @ W : b operand must be column-major
        w8_t = w8.t().contiguous()          # [K, N] row-major
        dx = _scaled_mm(g8, w8_t.t(), inv_g, inv_w, out_dtype=g.dtype)
        # dW = g^T @ x
        g8_t = g8.t().contiguous()
        dw = _scaled_mm(g8_t, x8.t().contiguous().t(), inv_g, inv_x,
                        out_dtype=g.dtype)
        db = g2.sum(0) if ctx.has_bias else None
        return dx.view(ctx.x_shape), dw, db


class Float8Linear(nn.Linear):
    """Drop-in nn.Linear running fp8 MFMA GEMMs with dynamic scaling."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not x.is_cuda:
            return super().forward(x)
        T = x.numel() // x.shape[-1]
        if T % 16 != 0:
            return super().forward(x)
        return _Fp8LinearFn.apply(x, self.weight, self.bias)

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "Float8Linear":
        m = cls.__new__(cls)
        nn.Module.__init__(m)
        m.in_features = lin.in_features
        m.out_features = lin.out_features
        m.weight = lin.weight
        m.bias = lin.bias
        return m


def apply_fp8_to_model(
    model: nn.Module,
    include: tuple[str, ...] = ("q_proj", "k_proj", "v_proj", "o_proj",
                                "gate_proj", "up_proj", "down_proj"),
    min_dim: int = 512,
) -> int:
    """Swap matching nn.Linear modules for Float8Linear. Returns swap count.
    (reference fp8.py apply_fp8_to_model: module-filter based swap)."""
    n = 0
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if (
                isinstance(child, nn.Linear)
                and not isinstance(child, Float8Linear)
                and child_name in include
                and child.in_features % 16 == 0 and child.out_features % 16 == 0
                and min(child.in_features, child.out_features) >= min_dim
            ):
                setattr(module, child_name, Float8Linear.from_linear(child))
                n += 1
    return n
