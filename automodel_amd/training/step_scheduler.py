"""Step scheduling: epochs, grad-accum batches, validation/checkpoint cadence.

Reference behavior: nemo_automodel/components/training/step_scheduler.py:56-320
(StepScheduler yields lists of grad-accum microbatches and exposes
``is_val_step`` / ``is_ckpt_step`` cadence properties, is a Stateful for
checkpoint resume, and reacts to SIGTERM).
"""

from __future__ import annotations

from typing import Any, Iterator


class StepScheduler:
    """Iterates a dataloader into grad-accumulation groups.

    Each ``__iter__`` pass over an epoch yields lists of ``grad_acc_steps``
    microbatches. ``step`` counts optimizer steps (global steps).
    """

    def __init__(
        self,
        grad_acc_steps: int = 1,
        ckpt_every_steps: int = 0,
        val_every_steps: int = 0,
        max_steps: int | None = None,
        num_epochs: int = 1,
        dataloader: Any = None,
    ):
        if grad_acc_steps < 1:
            raise ValueError("grad_acc_steps must be >= 1")
        self.grad_acc_steps = grad_acc_steps
        self.ckpt_every_steps = ckpt_every_steps
        self.val_every_steps = val_every_steps
        self.max_steps = max_steps
        self.num_epochs = num_epochs
        self.dataloader = dataloader
        self.step = 0
        self.epoch = 0
        self.sigterm_received = False

    # -- iteration ------------------------------------------------------------
    @property
    def epochs(self) -> Iterator[int]:
        while self.epoch < self.num_epochs and not self.finished:
            yield self.epoch
            self.epoch += 1

    @property
    def finished(self) -> bool:
        if self.sigterm_received:
            return True
        return self.max_steps is not None and self.step >= self.max_steps

    def __iter__(self) -> Iterator[list]:
        assert self.dataloader is not None, "StepScheduler needs a dataloader to iterate"
        batch_group: list = []
        for batch in self.dataloader:
            if self.finished:
                break
            batch_group.append(batch)
            if len(batch_group) == self.grad_acc_steps:
                self.step += 1
                yield batch_group
                batch_group = []
        # drop incomplete trailing group (parity with reference: only full
        # grad-accum groups become optimizer steps)

    # -- cadence --------------------------------------------------------------
    @property
    def is_val_step(self) -> bool:
        return self.val_every_steps > 0 and self.step % self.val_every_steps == 0

    @property
    def is_ckpt_step(self) -> bool:
        return self.ckpt_every_steps > 0 and self.step % self.ckpt_every_steps == 0

    # -- Stateful protocol (torch.distributed.checkpoint) ----------------------
    def state_dict(self) -> dict:
        return {"step": self.step, "epoch": self.epoch}

    def load_state_dict(self, state: dict) -> None:
        self.step = int(state["step"])
        self.epoch = int(state["epoch"])
