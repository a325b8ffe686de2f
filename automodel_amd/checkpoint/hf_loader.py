"""Load HF safetensors checkpoints into (possibly FSDP2-sharded) models.

Reference behavior: nemo_automodel/components/checkpoint/checkpointing.py:1228
(load_base_model: initial HF weights loaded into the already-sharded model).
Here each rank mmaps the safetensors shards and set_model_state_dict scatters
into DTensor shards; 288 GB HBM3E means full-tensor staging is fine far beyond
8B-class models on a single node.
"""

from __future__ import annotations

import json
import os

import torch

from automodel_amd.ops.rope import build_rope_cache


def _iter_safetensor_files(path: str):
    idx = os.path.join(path, "model.safetensors.index.json")
    if os.path.exists(idx):
        with open(idx) as f:
            index = json.load(f)
        for fn in sorted(set(index["weight_map"].values())):
            yield os.path.join(path, fn)
    else:
        for fn in sorted(os.listdir(path)):
            if fn.endswith(".safetensors"):
                yield os.path.join(path, fn)


def load_hf_state_dict(path: str) -> dict[str, torch.Tensor]:
    from safetensors.torch import load_file

    sd: dict[str, torch.Tensor] = {}
    for f in _iter_safetensor_files(path):
        sd.update(load_file(f))
    return sd


def load_hf_weights(model: torch.nn.Module, path: str, device=None, strict: bool = True) -> None:
    """Adapt HF keys via the model's state_dict_adapter (identity for llama)
    and load into the live (sharded or plain) model."""
    sd = load_hf_state_dict(path)
    adapter = getattr(model, "state_dict_adapter", None)
    if adapter is not None:
        sd = adapter.from_hf(sd)

    is_sharded = any(
        type(p).__name__ == "DTensor" for p in model.parameters()
    )
    if is_sharded:
        from torch.distributed.checkpoint.state_dict import (
            StateDictOptions,
            set_model_state_dict,
        )

        set_model_state_dict(
            model, sd,
            options=StateDictOptions(full_state_dict=True, strict=strict),
        )
    else:
        if device is not None and any(p.is_meta for p in model.parameters()):
            model.to_empty(device=device)
        missing, unexpected = model.load_state_dict(sd, strict=False, assign=False)
        real_missing = [m for m in missing if "rope_cos" not in m and "rope_sin" not in m]
        if strict and (real_missing or unexpected):
            raise RuntimeError(f"HF load mismatch: missing={real_missing} unexpected={unexpected}")
    # non-persistent rope buffers are not in checkpoints — rebuild on device
    cfg = getattr(model, "config", None)
    if cfg is not None and hasattr(model, "model") and hasattr(model.model, "rope_cos"):
        dev = next(model.parameters()).device
        cos, sin = build_rope_cache(cfg.head_dim, cfg.max_position_embeddings,
                                    cfg.rope_theta, cfg.rope_scaling, device=dev)
        model.model.rope_cos = cos
        model.model.rope_sin = sin
