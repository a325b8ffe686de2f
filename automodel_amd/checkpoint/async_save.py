"""Async checkpoint save: CPU staging + background writer thread.

Reference behavior: nemo_automodel/components/checkpoint/checkpointing.py:363
(async save with staging to CPU on a dedicated gloo process group;
maybe_wait_for_staging before the next optimizer step).
"""

from __future__ import annotations

import threading
from typing import Callable

import torch


class AsyncCheckpointWriter:
    """Stage tensors to CPU synchronously (fast D2H over PCIe/xGMI), then
    write to disk on a background thread. One outstanding save at a time."""

    def __init__(self):
        self._thread: threading.Thread | None = None

    def wait(self) -> None:
        """Block until any in-flight save finishes (call before the next
        optimizer step mutates the weights — reference train_ft.py:1251)."""
        if self._thread is not None:
            self._thread.join()
            self._thread = None

    def stage(self, state: dict) -> dict:
        def _stage(obj):
            if isinstance(obj, torch.Tensor):
                t = obj
                if hasattr(t, "to_local"):
                    try:
                        from torch.distributed.tensor import DTensor

                        if isinstance(t, DTensor):
                            t = t.to_local()
                    except ImportError:
                        pass
                return t.detach().to("cpu", non_blocking=True).clone()
            if isinstance(obj, dict):
                return {k: _stage(v) for k, v in obj.items()}
            if isinstance(obj, (list, tuple)):
                return type(obj)(_stage(v) for v in obj)
            return obj

        staged = _stage(state)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        return staged

    def save_async(self, state: dict, write_fn: Callable[[dict], None]) -> None:
        self.wait()
        staged = self.stage(state)
        self._thread = threading.Thread(target=write_fn, args=(staged,), daemon=True)
        self._thread.start()

    @property
    def in_flight(self) -> bool:
        return self._thread is not None and self._thread.is_alive()
