"""Kernel-patch ladder for the generic HF-transformers fallback path.

Reference behavior: nemo_automodel/_transformers/model_init.py:1431 (the
reference patches AutoModel instances with its optimized kernels — liger /
TE swaps — stepping down a ladder of "patch what matches, leave the rest").
MI355X-native equivalents:

  rung 1 — RMSNorm: any module whose class name ends in "RMSNorm" with a
           1-D weight and an eps attr is swapped for ops.rms_norm.RMSNorm
           (HIP kernel on gfx950, torch path on CPU), REUSING the original
           weight Parameter so optimizers/checkpoints see the same tensor.
  rung 2 — SwiGLU MLP: modules with gate_proj/up_proj/down_proj and a silu
           act_fn get their forward rebound to the fused HIP swiglu
           (one kernel instead of two elementwise launches).

Patches are shape/attribute-gated: anything that doesn't match exactly is
left untouched, so arbitrary architectures stay correct.
"""

from __future__ import annotations

import types

import torch
import torch.nn as nn

from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.swiglu import swiglu


def _is_hf_rmsnorm(mod: nn.Module) -> bool:
    if type(mod).__name__ == "RMSNorm" and isinstance(mod, RMSNorm):
        return False  # already ours
    if not type(mod).__name__.endswith("RMSNorm"):
        return False
    w = getattr(mod, "weight", None)
    eps = getattr(mod, "variance_epsilon", getattr(mod, "eps", None))
    return (isinstance(w, nn.Parameter) and w.ndim == 1 and eps is not None
            and len(list(mod.parameters())) == 1
            and len(list(mod.children())) == 0)


def _is_silu_gate_mlp(mod: nn.Module) -> bool:
    for name in ("gate_proj", "up_proj", "down_proj"):
        child = getattr(mod, name, None)
        if not isinstance(child, nn.Linear):
            return False
    act = getattr(mod, "act_fn", None)
    return (isinstance(act, nn.SiLU)
            or type(act).__name__ == "SiLUActivation"
            or getattr(act, "__name__", "") == "silu")


def _fused_mlp_forward(self, x):
    return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


def apply_kernel_patches(model: nn.Module) -> dict:
    """Walk the model, swap matching modules in place. Returns patch counts."""
    counts = {"rms_norm": 0, "swiglu_mlp": 0}
    for parent in model.modules():
        for name, child in list(parent.named_children()):
            if _is_hf_rmsnorm(child):
                eps = getattr(child, "variance_epsilon", None)
                if eps is None:
                    eps = getattr(child, "eps")
                new = RMSNorm(child.weight.shape[0], float(eps), backend="auto")
                new.weight = child.weight     # SAME Parameter object
                setattr(parent, name, new)
                counts["rms_norm"] += 1
            elif _is_silu_gate_mlp(child):
                child.forward = types.MethodType(_fused_mlp_forward, child)
                counts["swiglu_mlp"] += 1
    return counts
