"""LoRA / PEFT: LinearLoRA wrapper, wildcard module matching, merge.

Reference behavior: nemo_automodel/components/_peft/lora.py:45-570
(PeftConfig, LinearLoRA with frozen base + A/B adapters, wildcard
target_modules matching, apply_lora_to_linear_modules freezes the base model).
The fused SGMV HIP kernels (reference _peft/lora_kernel.py) are a later
optimization; the adapter math here is two small GEMMs riding hipBLASLt.
"""

from __future__ import annotations

import fnmatch
import math
from dataclasses import dataclass, field

import torch
import torch.nn as nn


@dataclass
class PeftConfig:
    target_modules: list[str] = field(
        default_factory=lambda: ["*q_proj", "*k_proj", "*v_proj", "*o_proj"]
    )
    dim: int = 8                      # rank
    alpha: float = 16.0
    dropout: float = 0.0
    use_dora: bool = False
    quantize_base: bool = False       # QLoRA: NF4-quantize matched base linears
    quant_block_size: int = 64

    @classmethod
    def from_config(cls, cfg) -> "PeftConfig":
        if isinstance(cfg, cls):
            return cfg
        d = dict(cfg.items()) if hasattr(cfg, "items") else dict(cfg)
        d.pop("_target_", None)
        return cls(**{k: (list(v) if k == "target_modules" else v) for k, v in d.items()})


class _FusedLoRAFn(torch.autograd.Function):
    """Fused delta = (x @ A^T) @ B^T * scale (csrc/lora.hip — one kernel,
    the [M, r] intermediate never hits HBM; reference _peft/lora_kernel.py
    lora_forward_kernel). Backward recomputes t with two skinny GEMMs."""

    @staticmethod
    def forward(ctx, x2, wa, wb, scale):
        from automodel_amd.ops._backend import hip_ops

        delta = hip_ops().lora_fused_fwd(x2, wa, wb, scale)
        ctx.save_for_backward(x2, wa, wb)
        ctx.scale = scale
        return delta

    @staticmethod
    def backward(ctx, g):
        x2, wa, wb = ctx.saved_tensors
        s = ctx.scale
        g = g.contiguous()
        t = x2 @ wa.t()                  # [M, r] recompute
        dt = (g @ wb) * s
        dx = dt.to(x2.dtype) @ wa
        da = dt.t().to(x2.dtype) @ x2
        db = (g.t() @ t.to(g.dtype)) * s
        return dx, da.to(wa.dtype), db.to(wb.dtype), None


class LinearLoRA(nn.Module):
    """y = base(x) + (dropout(x) @ A^T) @ B^T * (alpha / r); base frozen.

    use_dora: DoRA decomposition (reference _peft/lora.py:227) — a learned
    per-output magnitude rescales the direction of W + BA.
    """

    def __init__(self, base: nn.Module, dim: int, alpha: float, dropout: float = 0.0,
                 use_dora: bool = False):
        super().__init__()
        self.base = base
        self.dim = dim
        self.scale = alpha / dim
        self.use_dora = use_dora
        base_w = self._base_weight(materialize=use_dora)
        if base_w is not None:
            dtype, dev = base_w.dtype, base_w.device
        else:  # NF4 base: match the dtype the base computed in pre-quantization
            dtype = getattr(base, "compute_dtype", torch.float32)
            dev = next(iter(base.buffers())).device
        self.lora_A = nn.Linear(base.in_features, dim, bias=False, dtype=dtype, device=dev)
        self.lora_B = nn.Linear(dim, base.out_features, bias=False, dtype=dtype, device=dev)
        self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()
        if use_dora:
            with torch.no_grad():
                mag = base_w.float().norm(dim=1) if not base_w.is_meta \
                    else torch.ones(base.out_features)
            self.lora_magnitude = nn.Parameter(mag.to(dtype=dtype, device=dev))
        self.reset_lora_parameters()
        for p in self.base.parameters():
            p.requires_grad_(False)

    def _base_weight(self, materialize: bool = True):
        """The effective base weight. For an NF4Linear base (QLoRA) this is
        the dequantized matrix (only materialized when needed)."""
        w = getattr(self.base, "weight", None)
        if w is not None:
            return w
        return self.base.dequantized_weight() if materialize else None

    def reset_lora_parameters(self) -> None:
        if not self.lora_A.weight.is_meta:
            nn.init.kaiming_uniform_(self.lora_A.weight, a=math.sqrt(5))
            nn.init.zeros_(self.lora_B.weight)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.use_dora:
            w = self._base_weight() + (self.lora_B.weight @ self.lora_A.weight) * self.scale
            col_norm = w.float().norm(dim=1).clamp_min(1e-6).to(w.dtype)
            w = w * (self.lora_magnitude / col_norm).unsqueeze(1)
            y = torch.nn.functional.linear(x, w, self.base.bias)
            return y
        wa, wb = self.lora_A.weight, self.lora_B.weight
        if (x.is_cuda and x.dtype == torch.bfloat16 and not self.training_dropout()
                and wa.shape[0] in (32, 64) and wa.shape[1] % 64 == 0
                and wb.shape[0] % 64 == 0
                and (x.numel() // x.shape[-1]) % 8 == 0):
            x2 = x.reshape(-1, x.shape[-1]).contiguous()
            delta = _FusedLoRAFn.apply(x2, wa, wb, self.scale)
            return self.base(x) + delta.view(*x.shape[:-1], wb.shape[0])
        return self.base(x) + self.lora_B(self.lora_A(self.dropout(x))) * self.scale

    def training_dropout(self) -> bool:
        return self.training and not isinstance(self.dropout, nn.Identity)

    @torch.no_grad()
    def merge(self) -> nn.Linear:
        """Fold the adapter into the base weight and return the plain Linear.
        A quantized (NF4) base is dequantized first — merge de-quantizes."""
        delta = (self.lora_B.weight @ self.lora_A.weight) * self.scale
        if not isinstance(self.base, nn.Linear):
            w = self._base_weight()
            lin = nn.Linear(self.base.in_features, self.base.out_features,
                            bias=self.base.bias is not None,
                            dtype=w.dtype, device=w.device)
            lin.weight.copy_(w)
            if self.base.bias is not None:
                lin.bias.copy_(self.base.bias)
            self.base = lin
        self.base.weight += delta.to(self.base.weight.dtype)
        return self.base


def apply_lora_to_linear_modules(model: nn.Module, cfg) -> int:
    """Freeze the model; replace matching nn.Linear with LinearLoRA. Returns
    the number of adapted modules (reference lora.py:570)."""
    peft = PeftConfig.from_config(cfg)
    for p in model.parameters():
        p.requires_grad_(False)

    replaced = 0
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            full = f"{name}.{child_name}" if name else child_name
            if isinstance(child, nn.Linear) and not isinstance(child, LinearLoRA):
                if any(fnmatch.fnmatch(full, pat) for pat in peft.target_modules):
                    if peft.quantize_base:
                        from automodel_amd.quantization.nf4 import NF4Linear

                        child = NF4Linear(child, peft.quant_block_size)
                    setattr(module, child_name,
                            LinearLoRA(child, peft.dim, peft.alpha, peft.dropout,
                                       use_dora=peft.use_dora))
                    replaced += 1
    return replaced


def lora_state_dict(model: nn.Module) -> dict[str, torch.Tensor]:
    """Adapter-only state dict (reference checkpoint/addons.py PEFT saves)."""
    return {k: v for k, v in model.state_dict().items() if "lora_" in k}


def export_hf_peft_adapter(model: nn.Module, out_dir: str,
                           base_model_name: str = "") -> None:
    """Write the adapters in HF-PEFT layout (adapter_config.json +
    adapter_model.safetensors with `base_model.model.` key prefixes and
    .weight suffixes) so `peft.PeftModel.from_pretrained` can load them
    (reference checkpoint/addons.py PEFT saves are HF-PEFT compatible)."""
    import json
    import os

    from safetensors.torch import save_file

    os.makedirs(out_dir, exist_ok=True)
    sd = {}
    target_modules: set[str] = set()
    rank = alpha = None
    for name, module in model.named_modules():
        if not isinstance(module, LinearLoRA):
            continue
        rank, alpha = module.dim, module.scale * module.dim
        target_modules.add(name.rsplit(".", 1)[-1])
        sd[f"base_model.model.{name}.lora_A.weight"] = \
            module.lora_A.weight.detach().contiguous()
        sd[f"base_model.model.{name}.lora_B.weight"] = \
            module.lora_B.weight.detach().contiguous()
    save_file(sd, os.path.join(out_dir, "adapter_model.safetensors"),
              metadata={"format": "pt"})
    cfg = {
        "peft_type": "LORA",
        "base_model_name_or_path": base_model_name,
        "r": rank,
        "lora_alpha": alpha,
        "lora_dropout": 0.0,
        "target_modules": sorted(target_modules),
        "bias": "none",
        "task_type": "CAUSAL_LM",
    }
    with open(os.path.join(out_dir, "adapter_config.json"), "w") as f:
        json.dump(cfg, f, indent=1)
