"""Distributed SIGTERM handling: all ranks observe termination together.

Reference behavior: nemo_automodel/components/training/signal_handler.py:94-199
(DistributedSignalHandler all-gathers per-rank signal flags each check so every
rank exits in lockstep and a final checkpoint can be written).
"""

from __future__ import annotations

import signal
from types import FrameType

import torch
import torch.distributed as dist


class DistributedSignalHandler:
    def __init__(self, sig: int = signal.SIGTERM):
        self.sig = sig
        self._received = False
        self._prev_handler = None

    def __enter__(self) -> "DistributedSignalHandler":
        self._prev_handler = signal.getsignal(self.sig)
        signal.signal(self.sig, self._handle)
        return self

    def __exit__(self, *exc) -> None:
        if self._prev_handler is not None:
            signal.signal(self.sig, self._prev_handler)

    def _handle(self, signum: int, frame: FrameType | None) -> None:
        self._received = True

    def signals_received(self) -> bool:
        """True iff ANY rank received the signal (collective)."""
        local = self._received
        if dist.is_available() and dist.is_initialized():
            t = torch.tensor([1 if local else 0], dtype=torch.int64)
            if dist.get_backend() == "nccl" and torch.cuda.is_available():
                t = t.cuda()
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            return bool(t.item())
        return local
