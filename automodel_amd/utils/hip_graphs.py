"""Partial hipGraph capture of selected modules (launch-bound inner loops).

Reference behavior: nemo_automodel/components/cuda_graphs/partial.py:868
(PartialCudaGraphManager: scoped CUDA-graph capture of selected module
classes after the first eager step). On ROCm, torch.cuda.CUDAGraph IS a
hipGraph — capture replays the recorded HIP kernel launches with one
hipGraphLaunch, removing per-kernel launch latency for small static-shape
modules (e.g. MoE routers).

Constraints: captured modules must be static-shape and free of host syncs
inside forward (guide Guideline 9). Capture is forward-only (inference /
frozen modules) — training graphs need the full make_graphed_callables
machinery and are out of scope here.
"""

from __future__ import annotations

import torch
import torch.nn as nn


class GraphedForward(nn.Module):
    """Wraps a module; after `warmup` eager calls with a stable input shape,
    captures one hipGraph and replays it thereafter."""

    def __init__(self, module: nn.Module, warmup: int = 3):
        super().__init__()
        self.inner = module
        self.warmup = warmup
        self._calls = 0
        self._graph = None
        self._static_in: torch.Tensor | None = None
        self._static_out: torch.Tensor | None = None

    @torch.no_grad()
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not x.is_cuda or torch.is_grad_enabled():
            return self.inner(x)
        if self._graph is not None and x.shape == self._static_in.shape:
            self._static_in.copy_(x)
            self._graph.replay()
            return self._static_out
        self._calls += 1
        out = self.inner(x)
        if self._calls >= self.warmup:
            self._static_in = x.clone()
            self._graph = torch.cuda.CUDAGraph()
            torch.cuda.synchronize()
            with torch.cuda.graph(self._graph):
                self._static_out = self.inner(self._static_in)
            torch.cuda.synchronize()
        return out


def apply_partial_graphs(model: nn.Module, cls_names: tuple[str, ...],
                         warmup: int = 3) -> int:
    """Wrap every matching (frozen, static-shape) submodule. Returns count."""
    n = 0
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if type(child).__name__ in cls_names:
                setattr(module, child_name, GraphedForward(child, warmup))
                n += 1
    return n


def apply_training_graphs(model: nn.Module, cls_names: tuple[str, ...],
                          sample_input: torch.Tensor, warmup: int = 3) -> int:
    """hipGraph-capture matching submodules for TRAINING (forward+backward
    replay) via torch.cuda.make_graphed_callables — the reference's partial
    CUDA-graph manager equivalent for the train loop. ``sample_input`` must
    match the static per-module input shape/dtype/device. Returns count."""
    targets = []
    for name, module in model.named_modules():
        for child_name, child in module.named_children():
            if type(child).__name__ in cls_names:
                targets.append((module, child_name, child))
    if not targets:
        return 0
    graphed = torch.cuda.make_graphed_callables(
        tuple(t[2] for t in targets),
        tuple((sample_input.clone().requires_grad_(True),) for _ in targets),
        num_warmup_iters=warmup,
    )
    if not isinstance(graphed, tuple):
        graphed = (graphed,)
    for (module, child_name, _), g in zip(targets, graphed):
        setattr(module, child_name, g)
    return len(targets)
