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


def _save_ops() -> set:
    """Ops whose outputs are kept under selective-op AC: the GEMM-shaped and
    attention ops (expensive to recompute); everything elementwise/norm-like
    is recomputed (cheap, HBM-bound)."""
    ops = {
        torch.ops.aten.mm.default,
        torch.ops.aten.addmm.default,
        torch.ops.aten.bmm.default,
        torch.ops.aten._scaled_dot_product_flash_attention.default,
        torch.ops.aten._scaled_dot_product_efficient_attention.default,
    }
    try:
        ops.add(torch.ops.aten._scaled_mm.default)
    except AttributeError:
        pass
    try:  # in-tree flash forward: never recompute attention
        ops.add(torch.ops.amd_ops.flash_attn_fwd.default)
    except (AttributeError, RuntimeError):
        pass
    return ops


def _selective_op_checkpoint_fn():
    """torch.utils.checkpoint fn with a save-GEMMs/recompute-elementwise
    policy (reference activation_checkpointing.py selective-op mode)."""
    from torch.utils.checkpoint import (
        CheckpointPolicy,
        checkpoint,
        create_selective_checkpoint_contexts,
    )

    save = _save_ops()

    def policy(ctx, op, *args, **kwargs):
        return (CheckpointPolicy.MUST_SAVE if op in save
                else CheckpointPolicy.PREFER_RECOMPUTE)

    return partial(checkpoint, use_reentrant=False,
                   context_fn=partial(create_selective_checkpoint_contexts, policy))


def apply_ac(
    model: nn.Module,
    mode: str = "full",                      # full | selective | selective_ops
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
    targets = selective_cls_names if mode == "selective" else layer_cls_names
    count = 0

    def check_fn(m: nn.Module) -> bool:
        nonlocal count
        if type(m).__name__ in targets:
            count += 1
            return (count - 1) % every_n == 0
        return False

    if mode == "selective_ops":
        # full-layer wrap, but the policy keeps GEMM/attention outputs and
        # recomputes only the elementwise/norm tail — trades a fraction of
        # full-AC's memory saving for near-zero recompute FLOPs
        wrapper = partial(checkpoint_wrapper,
                          checkpoint_impl=CheckpointImpl.NO_REENTRANT,
                          checkpoint_fn=_selective_op_checkpoint_fn())
    else:
        wrapper = partial(checkpoint_wrapper,
                          checkpoint_impl=CheckpointImpl.NO_REENTRANT)
    apply_activation_checkpointing(model, checkpoint_wrapper_fn=wrapper,
                                   check_fn=check_fn)
    return model
