"""FSDP2 (fully_shard) sharding tuned for MI355X.

Reference behavior: nemo_automodel/components/distributed/parallelizer.py:1192
(apply_fsdp2_sharding_recursively: per-decoder-layer fully_shard, root
unsharded params, explicit prefetch) and fsdp2.py:85 (FSDP2Manager).

MI355X-first choices (SURVEY §7 Phase 1): with 288 GB HBM3E per GPU,
``reshard_after_forward=False`` is the default (params stay gathered between
forward and backward — xGMI all-gather happens once per step, not twice), and
the per-layer bucket IS the whole decoder layer (large collectives suit the
per-link-bound ring over 7x153 GB/s xGMI).
"""

from __future__ import annotations

import torch
import torch.nn as nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.fsdp import MixedPrecisionPolicy, fully_shard


def detect_decoder_layers(model: nn.Module) -> list[nn.Module]:
    """Generic transformer-block detection for models whose layer class is
    not in the known list (the HF-transformers fallback path): pick the
    largest nn.ModuleList whose children all share one type — that is the
    decoder stack in every HF causal LM (reference parallelizer.py's
    per-model strategies fall back to the same per-layer granularity)."""
    best: list[nn.Module] = []
    for m in model.modules():
        if isinstance(m, nn.ModuleList) and len(m) >= 2:
            kinds = {type(c) for c in m}
            if len(kinds) == 1 and len(m) > len(best):
                best = list(m)
    return best


def apply_fsdp(
    model: nn.Module,
    mesh: DeviceMesh,
    layer_cls_names: tuple[str, ...] = (
        "LlamaDecoderLayer", "MoEDecoderLayer", "VisionBlock",
        "GemmaDecoderLayer", "GptOssDecoderLayer", "NemotronDecoderLayer",
        "Glm4MoeDecoderLayer", "DeepseekV3DecoderLayer",
        "SiglipEncoderLayer", "ClipEncoderLayer", "PixtralLayer",
    ),
    param_dtype: torch.dtype = torch.bfloat16,
    reduce_dtype: torch.dtype = torch.float32,
    reshard_after_forward: bool = False,
) -> nn.Module:
    """Shard each decoder layer, then the root; wire backward prefetch.

    ``mesh`` may be 1-D (pure FSDP over dp_shard / dp_shard_cp) or 2-D
    (dp_replicate, dp_shard) for HSDP — fully_shard then replicates grads
    over the first axis with all-reduce (reference HSDP via dp_replicate).
    """
    mp = MixedPrecisionPolicy(param_dtype=param_dtype, reduce_dtype=reduce_dtype)
    layers = [
        m for m in model.modules() if type(m).__name__ in layer_cls_names
    ]
    if not layers:
        layers = detect_decoder_layers(model)
    for layer in layers:
        fully_shard(layer, mesh=mesh, mp_policy=mp, reshard_after_forward=reshard_after_forward)
    fully_shard(model, mesh=mesh, mp_policy=mp, reshard_after_forward=reshard_after_forward)

    # explicit prefetch: each layer prefetches the next (fwd) / previous (bwd)
    # at depth 2, mirroring the reference's DefaultParallelizationStrategy
    # (parallelizer.py:290-291); FSDP2 does implicit depth-1 already.
    for i, layer in enumerate(layers):
        if i + 2 < len(layers):
            layer.set_modules_to_forward_prefetch([layers[i + 1], layers[i + 2]])
        if i - 2 >= 0:
            layer.set_modules_to_backward_prefetch([layers[i - 1], layers[i - 2]])
    return model
