"""BaseRecipe: checkpoint-aware attribute tracking + mesh helpers.

Reference behavior: nemo_automodel/recipes/base_recipe.py:161-995
(__setattr__ auto-registers any attribute exposing state_dict/load_state_dict
as checkpointable; _dp_allreduce helpers; save/load_checkpoint orchestration).
"""

from __future__ import annotations

import os
from typing import Any

import torch
import torch.distributed as dist

from automodel_amd.config.loader import ConfigNode


def _is_stateful(obj: Any) -> bool:
    return callable(getattr(obj, "state_dict", None)) and callable(
        getattr(obj, "load_state_dict", None)
    )


class BaseRecipe:
    def __init__(self, cfg: ConfigNode):
        object.__setattr__(self, "_statefuls", {})
        self.cfg = cfg

    def __setattr__(self, name: str, value: Any) -> None:
        if _is_stateful(value) and not name.startswith("_"):
            self._statefuls[name] = value
        object.__setattr__(self, name, value)

    # -- distributed helpers ---------------------------------------------------
    @property
    def rank(self) -> int:
        return dist.get_rank() if dist.is_initialized() else 0

    @property
    def world_size(self) -> int:
        return dist.get_world_size() if dist.is_initialized() else 1

    def _dp_allreduce(self, t: torch.Tensor, group=None, op=dist.ReduceOp.SUM) -> torch.Tensor:
        """All-reduce over the DP group (reference base_recipe.py:825)."""
        if dist.is_initialized() and (group is not None or self.world_size > 1):
            dist.all_reduce(t, op=op, group=group)
        return t

    # -- checkpoint ------------------------------------------------------------
    def save_checkpoint(self, path: str) -> None:
        """Save every tracked Stateful. Model/optimizer states go through the
        Checkpointer when one is registered (attribute ``checkpointer``)."""
        ckpt = getattr(self, "checkpointer", None)
        os.makedirs(path, exist_ok=True)
        aux = {}
        for name, obj in self._statefuls.items():
            if name in ("model", "optimizer", "checkpointer"):
                continue
            aux[name] = obj.state_dict()
        if ckpt is not None:
            ckpt.save(path, model=getattr(self, "model", None),
                      optimizer=getattr(self, "optimizer", None), extra_state=aux,
                      rank=self.rank)
        else:
            if self.rank == 0:
                torch.save(aux, os.path.join(path, "aux_state.pt"))

    def load_checkpoint(self, path: str) -> None:
        ckpt = getattr(self, "checkpointer", None)
        if ckpt is not None:
            aux = ckpt.load(path, model=getattr(self, "model", None),
                            optimizer=getattr(self, "optimizer", None), rank=self.rank)
        else:
            aux_path = os.path.join(path, "aux_state.pt")
            aux = torch.load(aux_path, weights_only=True) if os.path.exists(aux_path) else {}
        for name, state in (aux or {}).items():
            if name in self._statefuls:
                self._statefuls[name].load_state_dict(state)
