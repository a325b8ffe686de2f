"""Quantization-aware training: straight-through fake-quant linears.

Reference behavior: nemo_automodel/components/quantization/qat.py:45-169
(QATConfig with quantizer_type "int8_dynact_int4weight" | "int4_weight_only",
prepare applied after model build, enable/disable fake-quant toggles for
delayed fake-quant during training). The reference defers to torchao's QAT
quantizers; torchao is not a dependency here, so the fake-quant math is
implemented directly: symmetric per-group int4 weights and per-token dynamic
int8 activations with an identity straight-through estimator
(w + (quant(w) - w).detach()), which runs as a handful of elementwise ops
fused by the HIP runtime and adds no GEMM-path changes — the quantized
forward still rides hipBLASLt in bf16.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


def fake_quant_per_group(w: torch.Tensor, n_bits: int = 4,
                         group_size: int = 32) -> torch.Tensor:
    """Symmetric per-group fake quantization along the last dim, STE gradient."""
    qmax = 2 ** (n_bits - 1) - 1
    orig_shape = w.shape
    assert w.shape[-1] % group_size == 0, (w.shape, group_size)
    g = w.reshape(*w.shape[:-1], -1, group_size)
    scale = g.abs().amax(dim=-1, keepdim=True).clamp_min(1e-8) / qmax
    q = (g / scale).round().clamp(-qmax - 1, qmax) * scale
    q = q.reshape(orig_shape)
    return w + (q - w).detach()


def fake_quant_per_token(x: torch.Tensor, n_bits: int = 8) -> torch.Tensor:
    """Symmetric per-token (last-dim) dynamic fake quantization, STE gradient."""
    qmax = 2 ** (n_bits - 1) - 1
    scale = x.abs().amax(dim=-1, keepdim=True).clamp_min(1e-8) / qmax
    q = (x / scale).round().clamp(-qmax - 1, qmax) * scale
    return x + (q - x).detach()


@dataclass
class QATConfig:
    """quantizer_type: "int8_dynact_int4weight" (8-bit dynamic activations +
    4-bit grouped weights) or "int4_weight_only". delay_steps: train in full
    precision first, enable fake-quant at this step (reference delayed
    fake-quant toggles)."""

    quantizer_type: str = "int8_dynact_int4weight"
    group_size: int = 32
    delay_steps: int = 0
    skip_modules: tuple = ("lm_head", "lora_")  # never fake-quant adapters

    def __post_init__(self):
        valid = ("int8_dynact_int4weight", "int4_weight_only")
        if self.quantizer_type not in valid:
            raise ValueError(f"quantizer_type must be one of {valid}")

    @classmethod
    def from_config(cls, cfg) -> "QATConfig":
        if isinstance(cfg, cls):
            return cfg
        d = dict(cfg.items()) if hasattr(cfg, "items") else dict(cfg)
        d.pop("_target_", None)
        d.pop("enabled", None)
        if "skip_modules" in d:
            d["skip_modules"] = tuple(d["skip_modules"])
        return cls(**d)


class QATLinear(nn.Linear):
    """nn.Linear with fake-quantized weight (and optionally activations).
    `fake_quant_enabled` mirrors the reference's enable/disable toggles."""

    def __init__(self, base: nn.Linear, qcfg: QATConfig):
        factory = {"device": base.weight.device, "dtype": base.weight.dtype}
        super().__init__(base.in_features, base.out_features,
                         bias=base.bias is not None, **factory)
        with torch.no_grad():
            self.weight.copy_(base.weight)
            if base.bias is not None:
                self.bias.copy_(base.bias)
        self.qcfg = qcfg
        self.fake_quant_enabled = qcfg.delay_steps == 0

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.fake_quant_enabled:
            return F.linear(x, self.weight, self.bias)
        w = fake_quant_per_group(self.weight, 4, self.qcfg.group_size)
        if self.qcfg.quantizer_type == "int8_dynact_int4weight":
            x = fake_quant_per_token(x, 8)
        return F.linear(x, w, self.bias)

    @torch.no_grad()
    def convert(self) -> tuple[torch.Tensor, torch.Tensor]:
        """-> (int8 codes [out, in], fp32 scales [out, in/group]) for export."""
        qmax = 7
        g = self.weight.float().reshape(self.out_features, -1, self.qcfg.group_size)
        scale = g.abs().amax(dim=-1).clamp_min(1e-8) / qmax
        codes = (g / scale[..., None]).round().clamp(-8, 7).to(torch.int8)
        return codes.reshape(self.out_features, self.in_features), scale


def prepare_qat(model: nn.Module, cfg) -> int:
    """Swap every eligible nn.Linear for QATLinear. Returns count swapped."""
    qcfg = QATConfig.from_config(cfg)
    count = 0
    for name, parent in list(model.named_modules()):
        for child_name, child in list(parent.named_children()):
            full = f"{name}.{child_name}" if name else child_name
            if type(child) is nn.Linear and not any(s in full for s in qcfg.skip_modules):
                setattr(parent, child_name, QATLinear(child, qcfg))
                count += 1
    return count


def set_fake_quant(model: nn.Module, enabled: bool) -> None:
    for m in model.modules():
        if isinstance(m, QATLinear):
            m.fake_quant_enabled = enabled


def maybe_enable_delayed_fake_quant(model: nn.Module, step: int) -> None:
    """Call once per train step: turns fake-quant on when step reaches the
    configured delay (reference delayed fake-quant)."""
    for m in model.modules():
        if isinstance(m, QATLinear) and not m.fake_quant_enabled \
                and step >= m.qcfg.delay_steps:
            m.fake_quant_enabled = True
