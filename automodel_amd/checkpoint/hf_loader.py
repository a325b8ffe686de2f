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


def load_hf_weights_streaming(model: torch.nn.Module, path: str,
                              strict: bool = True) -> None:
    """Shard-by-shard HF load into a DTensor-sharded model WITHOUT ever
    materializing the full state dict per rank (reference
    checkpointing.py:1228 load_base_model — DCP-from-HF streaming is what
    makes 671B-class from_pretrained possible).

    Peak host memory is bounded by one safetensors shard file plus any
    adapter keys still waiting for partner tensors (fused qkv / stacked MoE
    experts — HF checkpoints keep a layer's tensors in one shard, so the
    leftover set stays per-layer-sized). Each DTensor param takes only its
    local slice via distribute_tensor(src_data_rank=None) — no collective,
    no full-tensor GPU staging."""
    from torch.distributed.tensor import DTensor, distribute_tensor

    adapter = getattr(model, "state_dict_adapter", None)
    params: dict[str, torch.Tensor] = dict(model.named_parameters())
    params.update({k: v for k, v in model.named_buffers()
                   if "rope_cos" not in k and "rope_sin" not in k})
    done: set[str] = set()
    pending: dict[str, torch.Tensor] = {}

    def _assign(name: str, full: torch.Tensor) -> bool:
        p = params[name]
        if tuple(full.shape) != tuple(p.shape):
            # adapter output from a partial pending set (e.g. a stacked MoE
            # tensor with experts still missing) — wait for more files
            return False
        with torch.no_grad():
            if isinstance(p, DTensor):
                dt = distribute_tensor(full.to(p.dtype), p.device_mesh, p.placements,
                                       src_data_rank=None)
                p.detach().to_local().copy_(dt.to_local())
            else:
                p.copy_(full.to(p.dtype))
        done.add(name)
        return True

    key_targets = getattr(adapter, "hf_key_targets", None)
    for f in _iter_safetensor_files(path):
        from safetensors.torch import load_file

        pending.update(load_file(f))
        mapped = adapter.from_hf(dict(pending)) if adapter is not None else pending
        for k in list(mapped):
            if k in params and k not in done:
                _assign(k, mapped[k])
        # free consumed source keys: identity keys once assigned; adapter-
        # combined keys once every target they feed is assigned (adapters
        # advertise the mapping via hf_key_targets; without it the key is
        # kept — correct, just less memory-frugal)
        for k in list(pending):
            if k in done:
                del pending[k]
            elif key_targets is not None:
                tgts = key_targets(k)
                if tgts and all(t in done for t in tgts):
                    del pending[k]

    missing = [k for k in params if k not in done]
    if strict and missing:
        raise RuntimeError(f"HF streaming load: params never matched: {missing[:8]}"
                           f" (+{max(0, len(missing) - 8)} more)")


def load_hf_weights(model: torch.nn.Module, path: str, device=None, strict: bool = True) -> None:
    """Adapt HF keys via the model's state_dict_adapter (identity for llama)
    and load into the live (sharded or plain) model."""
    is_sharded = any(
        type(p).__name__ == "DTensor" for p in model.parameters()
    )
    if is_sharded:
        # sharded model: stream shard files, never the full state dict
        load_hf_weights_streaming(model, path, strict=strict)
    else:
        sd = load_hf_state_dict(path)
        adapter = getattr(model, "state_dict_adapter", None)
        if adapter is not None:
            sd = adapter.from_hf(sd)
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
