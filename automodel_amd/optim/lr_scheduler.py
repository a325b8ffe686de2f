"""Warmup + cosine/linear/constant LR schedule (Stateful).

Reference behavior: OptimizerParamScheduler in
nemo_automodel/components/optim/scheduler.py (warmup then decay to min_lr).
"""

from __future__ import annotations

import math

import torch


class WarmupDecayLR:
    def __init__(
        self,
        optimizer: torch.optim.Optimizer,
        warmup_steps: int = 0,
        total_steps: int | None = None,
        decay: str = "cosine",           # cosine | linear | constant
        min_lr_ratio: float = 0.0,
    ):
        self.optimizer = optimizer
        self.warmup_steps = warmup_steps
        self.total_steps = total_steps
        self.decay = decay
        self.min_lr_ratio = min_lr_ratio
        self.base_lrs = [g["lr"] for g in optimizer.param_groups]
        self.step_count = 0
        self._apply()

    def _factor(self) -> float:
        t = self.step_count
        if self.warmup_steps > 0 and t < self.warmup_steps:
            return (t + 1) / self.warmup_steps
        if self.decay == "constant" or not self.total_steps:
            return 1.0
        prog = min(1.0, (t - self.warmup_steps) / max(1, self.total_steps - self.warmup_steps))
        if self.decay == "cosine":
            f = 0.5 * (1 + math.cos(math.pi * prog))
        else:  # linear
            f = 1.0 - prog
        return self.min_lr_ratio + (1 - self.min_lr_ratio) * f

    def _apply(self) -> None:
        f = self._factor()
        for g, base in zip(self.optimizer.param_groups, self.base_lrs):
            g["lr"] = base * f

    def step(self) -> None:
        self.step_count += 1
        self._apply()

    def get_last_lr(self) -> list[float]:
        return [g["lr"] for g in self.optimizer.param_groups]

    def state_dict(self) -> dict:
        return {"step_count": self.step_count, "base_lrs": self.base_lrs}

    def load_state_dict(self, state: dict) -> None:
        self.step_count = state["step_count"]
        self.base_lrs = state["base_lrs"]
        self._apply()
