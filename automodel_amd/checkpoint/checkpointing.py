"""Checkpointer: DCP sharded save/load + consolidated HF-safetensors export.

Reference behavior: nemo_automodel/components/checkpoint/checkpointing.py:479
(Checkpointer.save_model/load_model: torch.distributed.checkpoint sharded
save, HF-safetensors consolidation modes, retention policy lifecycle.py).
"""

from __future__ import annotations

import os
import shutil

import torch
import torch.distributed as dist


def _sd_options(full: bool = False):
    from torch.distributed.checkpoint.state_dict import StateDictOptions

    return StateDictOptions(full_state_dict=full, cpu_offload=full)


class Checkpointer:
    def __init__(
        self,
        checkpoint_dir: str = "checkpoints",
        model_save_format: str = "safetensors",
        save_consolidated: bool = False,
        keep_last_n: int | None = None,
        keep_top_k: int | None = None,
        metric_higher_is_better: bool = False,
        async_save: bool = False,
    ):
        self.checkpoint_dir = checkpoint_dir
        self.model_save_format = model_save_format
        self.save_consolidated = save_consolidated
        self.keep_last_n = keep_last_n
        self.keep_top_k = keep_top_k             # by recorded metric (lifecycle.py)
        self.metric_higher_is_better = metric_higher_is_better
        self._step_metrics: dict = {}
        self.async_save = async_save
        self._async_writer = None
        if async_save:
            from automodel_amd.checkpoint.async_save import AsyncCheckpointWriter

            self._async_writer = AsyncCheckpointWriter()

    def maybe_wait_for_staging(self) -> None:
        """Block before the optimizer step mutates weights while an async
        save is still staging (reference train_ft.py:1251)."""
        if self._async_writer is not None:
            self._async_writer.wait()
        if getattr(self, "_async_future", None) is not None:
            self._async_future.result()
            self._async_future = None

    # ---------------------------------------------------------------- save
    def save(self, path: str, model=None, optimizer=None, extra_state: dict | None = None,
             rank: int = 0) -> None:
        import torch.distributed.checkpoint as dcp
        from torch.distributed.checkpoint.state_dict import (
            get_model_state_dict,
            get_optimizer_state_dict,
        )

        os.makedirs(path, exist_ok=True)
        state: dict = {}
        if model is not None:
            state["model"] = get_model_state_dict(model)
        if optimizer is not None:
            state["optimizer"] = get_optimizer_state_dict(model, optimizer)
        if state:
            if self.async_save:
                # staging happens inside async_save before it returns;
                # the write completes on a background thread. The next
                # optimizer step calls maybe_wait_for_staging().
                if getattr(self, "_async_future", None) is not None:
                    self._async_future.result()
                self._async_future = dcp.async_save(
                    state, checkpoint_id=os.path.join(path, "dcp"))
            else:
                dcp.save(state, checkpoint_id=os.path.join(path, "dcp"))
        if extra_state and rank == 0:
            torch.save(extra_state, os.path.join(path, "aux_state.pt"))
        if self.save_consolidated and model is not None:
            self.export_hf_safetensors(model, os.path.join(path, "hf"), rank=rank)
        if rank == 0:
            self._apply_retention()
        if dist.is_initialized():
            dist.barrier()

    def load(self, path: str, model=None, optimizer=None, rank: int = 0) -> dict:
        import torch.distributed.checkpoint as dcp
        from torch.distributed.checkpoint.state_dict import (
            get_model_state_dict,
            get_optimizer_state_dict,
            set_model_state_dict,
            set_optimizer_state_dict,
        )

        state: dict = {}
        if model is not None:
            state["model"] = get_model_state_dict(model)
        if optimizer is not None:
            state["optimizer"] = get_optimizer_state_dict(model, optimizer)
        if state:
            dcp.load(state, checkpoint_id=os.path.join(path, "dcp"))
            if model is not None:
                set_model_state_dict(model, state["model"])
            if optimizer is not None:
                set_optimizer_state_dict(model, optimizer, state["optimizer"])
        aux_path = os.path.join(path, "aux_state.pt")
        if os.path.exists(aux_path):
            # aux state is restricted to tensors/primitives (see StatefulRNG
            # et al.) so a tampered checkpoint dir cannot execute code on load
            return torch.load(aux_path, weights_only=True)
        return {}

    # ------------------------------------------------- consolidated HF export
    def export_hf_safetensors(self, model, out_dir: str, rank: int = 0) -> None:
        """Gather full state dict and write HF-layout safetensors on rank 0."""
        from torch.distributed.checkpoint.state_dict import get_model_state_dict

        full_sd = get_model_state_dict(model, options=_sd_options(full=True))
        if rank != 0:
            return
        from safetensors.torch import save_file

        os.makedirs(out_dir, exist_ok=True)
        self._export_config_json(model, out_dir)
        adapter = getattr(model, "state_dict_adapter", None)
        if adapter is not None:
            full_sd = adapter.to_hf(full_sd)
        full_sd = {k: v.contiguous() for k, v in full_sd.items() if isinstance(v, torch.Tensor)}
        # Drop only EXACT tied aliases (same data_ptr/shape/stride/dtype — the
        # tied-lm_head case; HF convention omits it and loaders re-tie from
        # config). Distinct views of one storage (fused-qkv splits, stacked
        # MoE expert slices from to_hf adapters) are real weights: keep them,
        # cloning so safetensors accepts the shared storage.
        seen: dict[tuple, str] = {}
        for k in list(full_sd):
            t = full_sd[k]
            key = (t.data_ptr(), tuple(t.shape), tuple(t.stride()), t.dtype)
            if key in seen:
                del full_sd[k]
            else:
                seen[key] = k
        kept_storages: set[int] = set()
        for k, t in full_sd.items():
            sp = t.untyped_storage().data_ptr()
            oversized = t.untyped_storage().nbytes() != t.numel() * t.element_size()
            if sp in kept_storages or oversized:
                full_sd[k] = t.clone()
            else:
                kept_storages.add(sp)
        total_bytes = sum(v.numel() * v.element_size() for v in full_sd.values())
        max_shard = 4 * 2**30
        if total_bytes <= max_shard:
            save_file(full_sd, os.path.join(out_dir, "model.safetensors"),
                      metadata={"format": "pt"})
            return
        # shard by size + write HF index (reference consolidate_hf_safetensors)
        import json as _json

        shards: list[dict] = [{}]
        sizes = [0]
        for k, v in full_sd.items():
            b = v.numel() * v.element_size()
            if sizes[-1] + b > max_shard and shards[-1]:
                shards.append({})
                sizes.append(0)
            shards[-1][k] = v
            sizes[-1] += b
        n = len(shards)
        weight_map = {}
        for i, shard in enumerate(shards):
            fn = f"model-{i+1:05d}-of-{n:05d}.safetensors"
            save_file(shard, os.path.join(out_dir, fn), metadata={"format": "pt"})
            for k in shard:
                weight_map[k] = fn
        with open(os.path.join(out_dir, "model.safetensors.index.json"), "w") as f:
            _json.dump({"metadata": {"total_size": total_bytes},
                        "weight_map": weight_map}, f)

    @staticmethod
    def _export_config_json(model, out_dir: str) -> None:
        """Write an HF-compatible config.json next to the weights so the
        export round-trips through pretrained_path loading."""
        import dataclasses
        import json as _json

        cfg = getattr(model, "config", None)
        if cfg is None or not dataclasses.is_dataclass(cfg):
            return
        d = {}
        for f in dataclasses.fields(cfg):
            v = getattr(cfg, f.name)
            if dataclasses.is_dataclass(v):        # nested (moe/text/vision)
                d[f.name] = dataclasses.asdict(v)
            elif isinstance(v, tuple):
                d[f.name] = list(v)
            elif isinstance(v, (int, float, str, bool, type(None), list, dict)):
                d[f.name] = v
        d["architectures"] = [type(model).__name__]
        with open(os.path.join(out_dir, "config.json"), "w") as fh:
            _json.dump(d, fh, indent=1)

    # ------------------------------------------------------------- retention
    def record_metric(self, step: int, value: float) -> None:
        """Associate a validation metric with a saved step for keep_top_k
        retention (reference checkpoint/lifecycle.py keep-top-k-by-metric)."""
        self._step_metrics[step] = float(value)

    def _apply_retention(self) -> None:
        if not os.path.isdir(self.checkpoint_dir):
            return
        steps = sorted(
            (d for d in os.listdir(self.checkpoint_dir) if d.startswith("step_")),
            key=lambda d: int(d.split("_")[1]),
        )
        drop: set = set()
        if self.keep_last_n:
            drop.update(steps[: -self.keep_last_n])
        if self.keep_top_k:
            scored = [d for d in steps if int(d.split("_")[1]) in self._step_metrics]
            ranked = sorted(
                scored,
                key=lambda d: self._step_metrics[int(d.split("_")[1])],
                reverse=self.metric_higher_is_better,
            )
            keep = set(ranked[: self.keep_top_k]) | ({steps[-1]} if steps else set())
            drop.update(d for d in scored if d not in keep)
            if self.keep_last_n:          # keep_last_n still protects the tail
                drop.difference_update(steps[-self.keep_last_n:])
        for d in drop:
            shutil.rmtree(os.path.join(self.checkpoint_dir, d), ignore_errors=True)
