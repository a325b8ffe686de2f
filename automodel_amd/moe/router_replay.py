"""Rollout Routing Replay (R3): record gate top-k decisions, replay later.

Reference behavior: nemo_automodel/components/moe/router_replay.py (record
the routing made during an RL rollout and replay the SAME expert choices
during the training forward so the policy gradient sees the rollout's
compute graph; probabilities are recomputed against the current weights).

Context-manager based (same pattern as the varlen / kv-cache contexts):
Gate.forward consults the active recorder — "record" appends each call's
indices, "replay" pops them back in call order.
"""

from __future__ import annotations

from contextlib import contextmanager

import torch

_ACTIVE: "RouterReplay | None" = None


class RouterReplay:
    def __init__(self):
        self.mode = "off"            # off | record | replay
        self.records: list[torch.Tensor] = []
        self._cursor = 0

    def start_record(self) -> None:
        self.mode = "record"
        self.records = []

    def start_replay(self) -> None:
        assert self.records, "nothing recorded"
        self.mode = "replay"
        self._cursor = 0

    def stop(self) -> None:
        self.mode = "off"

    # hooks used by Gate.forward -------------------------------------------
    def record(self, indices: torch.Tensor) -> None:
        self.records.append(indices.detach().clone())

    def next_replay(self) -> torch.Tensor:
        assert self._cursor < len(self.records), \
            "replay exhausted: more gate calls than were recorded"
        out = self.records[self._cursor]
        self._cursor += 1
        return out


@contextmanager
def router_replay_context(rr: RouterReplay):
    global _ACTIVE
    prev = _ACTIVE
    _ACTIVE = rr
    try:
        yield rr
    finally:
        _ACTIVE = prev


def active_router_replay() -> "RouterReplay | None":
    return _ACTIVE
