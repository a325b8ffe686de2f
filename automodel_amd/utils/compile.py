"""Regional torch.compile utilities.

Reference behavior: nemo_automodel/components/utils/compile_utils.py +
parallelizer.py:1107 (regional/per-layer compile). On ROCm, torch.compile
lowers through Triton which this framework does not ship — so regional
compile here targets the DYNAMO+eager backend ("aot_eager" by default) for
graph-level fusions, or any caller-specified backend. Per-layer compilation
keeps recompiles bounded and composes with FSDP2 (compile the decoder
layers, never the sharded root).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch.nn as nn


@dataclass
class CompileConfig:
    enabled: bool = False
    backend: str = "aot_eager"     # ROCm-safe default (no Triton dependency)
    mode: str | None = None
    fullgraph: bool = False
    regional: bool = True          # compile per decoder layer (not the root)

    @classmethod
    def from_config(cls, cfg) -> "CompileConfig":
        if isinstance(cfg, cls):
            return cfg
        d = dict(cfg.items()) if hasattr(cfg, "items") else dict(cfg or {})
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


def apply_compile(model: nn.Module, cfg: CompileConfig | None = None) -> int:
    """Compile each decoder layer in place (regional) or the whole model.
    Returns the number of compiled modules."""
    import torch

    cfg = cfg or CompileConfig(enabled=True)
    if not cfg.enabled:
        return 0
    kwargs = dict(backend=cfg.backend, fullgraph=cfg.fullgraph)
    if cfg.mode:
        kwargs["mode"] = cfg.mode
    if not cfg.regional:
        model.forward = torch.compile(model.forward, **kwargs)
        return 1
    from automodel_amd.parallel.fsdp import detect_decoder_layers

    n = 0
    for layer in detect_decoder_layers(model):
        layer.forward = torch.compile(layer.forward, **kwargs)
        n += 1
    return n
