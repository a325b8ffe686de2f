"""Activation checkpointing: full-layer and selective-module recompute.

Reference behavior: nemo_automodel/components/distributed/
activation_checkpointing.py:107-700 (full-layer AC, submodule AC limited to
attn/mlp, selective-op AC). On MI355X full AC is rarely needed for 8B-class
models (288 GB HBM3E) — it exists for the 70B+/long-context regimes.
"""

from __future__ import annotations

from functools import partial

import torch
import torch.nn as nn
from torch.distributed.algorithms._checkpoint.checkpoint_wrapper import (
    CheckpointImpl,
    apply_activation_checkpointing,
    checkpoint_wrapper,
)


def apply_ac(
    model: nn.Module,
    mode: str = "full",                      # full | selective
    layer_cls_names: tuple[str, ...] = (
        "LlamaDecoderLayer", "MoEDecoderLayer", "GemmaDecoderLayer",
        "GptOssDecoderLayer", "NemotronDecoderLayer", "Glm4MoeDecoderLayer",
        "DeepseekV3DecoderLayer",
    ),
    selective_cls_names: tuple[str, ...] = ("LlamaAttention", "LlamaMLP", "MoE"),
    every_n: int = 1,
) -> nn.Module:
    """Wrap decoder layers (mode=full) or attn/mlp submodules (selective) in
    non-reentrant torch.utils.checkpoint. ``every_n`` checkpoints every n-th
    matching module (partial AC)."""
    targets = layer_cls_names if mode == "full" else selective_cls_names
    count = 0

    def check_fn(m: nn.Module) -> bool:
        nonlocal count
        if type(m).__name__ in targets:
            count += 1
            return (count - 1) % every_n == 0
        return False

    apply_activation_checkpointing(
        model,
        checkpoint_wrapper_fn=partial(
            checkpoint_wrapper, checkpoint_impl=CheckpointImpl.NO_REENTRANT
        ),
        check_fn=check_fn,
    )
    return model
