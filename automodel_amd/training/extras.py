"""Training extras: EMA weights, NEFTune noise, stepped GC, memory monitor.

Reference behavior: nemo_automodel/components/training/{ema,neftune,
garbage_collection}.py and utils/ memory reporting.
"""

from __future__ import annotations

import gc

import torch
import torch.nn as nn


class EMA:
    """Exponential moving average of model parameters (reference training/ema.py)."""

    def __init__(self, model: nn.Module, decay: float = 0.999):
        self.decay = decay
        self.shadow = {
            name: p.detach().clone().float()
            for name, p in model.named_parameters() if p.requires_grad
        }

    @torch.no_grad()
    def update(self, model: nn.Module) -> None:
        for name, p in model.named_parameters():
            if name in self.shadow:
                self.shadow[name].mul_(self.decay).add_(p.detach().float(),
                                                        alpha=1 - self.decay)

    @torch.no_grad()
    def copy_to(self, model: nn.Module) -> None:
        for name, p in model.named_parameters():
            if name in self.shadow:
                p.copy_(self.shadow[name].to(p.dtype))

    def state_dict(self):
        return self.shadow

    def load_state_dict(self, state):
        self.shadow = state


def apply_neftune(embedding: nn.Embedding, alpha: float = 5.0) -> nn.Embedding:
    """NEFTune: uniform noise on embeddings during training
    (reference training/neftune.py)."""

    def hook(module, inputs, output):
        if module.training:
            dims = output.shape[-2] * output.shape[-1]
            eps = alpha / dims**0.5
            output = output + torch.empty_like(output).uniform_(-eps, eps)
        return output

    embedding.register_forward_hook(hook)
    return embedding


class SteppedGarbageCollector:
    """Disable automatic gc; collect on a fixed step cadence so all ranks
    pause together (reference training/garbage_collection.py)."""

    def __init__(self, every_steps: int = 100):
        self.every_steps = every_steps
        gc.disable()
        gc.collect()

    def maybe_collect(self, step: int) -> None:
        if self.every_steps > 0 and step % self.every_steps == 0:
            gc.collect()


def memory_stats() -> dict:
    if not torch.cuda.is_available():
        return {"allocated_gb": 0.0, "reserved_gb": 0.0}
    return {
        "allocated_gb": torch.cuda.memory_allocated() / 2**30,
        "max_allocated_gb": torch.cuda.max_memory_allocated() / 2**30,
        "reserved_gb": torch.cuda.memory_reserved() / 2**30,
    }


@torch.no_grad()
def repair_embedding_rows(embedding: nn.Embedding, reference_rows=None,
                          std: float = 0.02) -> int:
    """Re-init embedding rows containing NaN/Inf (reference
    training/embedding_row_repair.py — broken rows appear in some released
    checkpoints). Returns the number of rows repaired."""
    w = embedding.weight
    bad = ~torch.isfinite(w).all(dim=-1)
    n = int(bad.sum())
    if n:
        if reference_rows is not None:
            w[bad] = reference_rows.to(w.dtype)
        else:
            w[bad] = torch.randn(n, w.shape[1], device=w.device,
                                 dtype=w.dtype) * std
    return n
