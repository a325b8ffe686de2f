"""Checkpointable RNG state across python/numpy/torch, rank-offset.

Reference behavior: nemo_automodel/components/training/rng.py:121-200
(StatefulRNG seeds all three RNGs, optionally offsets by rank, and is a
Stateful; ScopedRNG is a context manager that restores prior state).
"""

from __future__ import annotations

import random

import numpy as np
import torch


class StatefulRNG:
    def __init__(self, seed: int = 42, ranked: bool = False):
        self.seed = seed
        self.ranked = ranked
        offset = 0
        if ranked and torch.distributed.is_available() and torch.distributed.is_initialized():
            offset = torch.distributed.get_rank()
        self._seed_all(seed + offset)

    @staticmethod
    def _seed_all(seed: int) -> None:
        random.seed(seed)
        np.random.seed(seed % (2**32))
        torch.manual_seed(seed)
        if torch.cuda.is_available():
            torch.cuda.manual_seed_all(seed)

    def state_dict(self) -> dict:
        # Plain tensors/primitives only, so checkpoints load with
        # torch.load(weights_only=True) (no arbitrary pickled objects).
        py = random.getstate()
        np_name, np_keys, np_pos, np_has_gauss, np_gauss = np.random.get_state()
        state = {
            "python": [py[0], list(py[1]), py[2]],
            "numpy": [np_name, torch.from_numpy(np_keys.copy()),
                      int(np_pos), int(np_has_gauss), float(np_gauss)],
            "torch": torch.get_rng_state(),
        }
        if torch.cuda.is_available():
            state["cuda"] = torch.cuda.get_rng_state_all()
        return state

    def load_state_dict(self, state: dict) -> None:
        py = state["python"]
        random.setstate((py[0], tuple(py[1]), py[2]))
        n = state["numpy"]
        np.random.set_state((n[0], n[1].numpy().astype(np.uint32), n[2], n[3], n[4]))
        torch.set_rng_state(torch.as_tensor(state["torch"], dtype=torch.uint8, device="cpu"))
        if "cuda" in state and torch.cuda.is_available():
            torch.cuda.set_rng_state_all(state["cuda"])


class ScopedRNG:
    """Temporarily reseed; restores previous RNG state on exit."""

    def __init__(self, seed: int):
        self.seed = seed
        self._saved: dict | None = None

    def __enter__(self) -> "ScopedRNG":
        self._saved = {
            "python": random.getstate(),
            "numpy": np.random.get_state(),
            "torch": torch.get_rng_state(),
        }
        StatefulRNG._seed_all(self.seed)
        return self

    def __exit__(self, *exc) -> None:
        assert self._saved is not None
        random.setstate(self._saved["python"])
        np.random.set_state(self._saved["numpy"])
        torch.set_rng_state(self._saved["torch"])
