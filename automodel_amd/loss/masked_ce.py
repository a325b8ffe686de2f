"""Masked token-sum cross entropy (logits path).

Reference behavior: nemo_automodel/components/loss/masked_ce.py:21-89
(token-sum CE with fp32 upcast; ignore_index masking; reduction left to the
recipe which divides by the DP-global number of label tokens).
"""

from __future__ import annotations

import torch


class MaskedCrossEntropy(torch.nn.Module):
    def __init__(self, ignore_index: int = -100, fp32_upcast: bool = True):
        super().__init__()
        self.ignore_index = ignore_index
        self.fp32_upcast = fp32_upcast

    def forward(self, logits: torch.Tensor, labels: torch.Tensor, mask: torch.Tensor | None = None) -> torch.Tensor:
        """logits [*, V], labels [*]; returns SUM of per-token losses."""
        logits = logits.reshape(-1, logits.shape[-1])
        labels = labels.reshape(-1)
        if mask is not None:
            labels = labels.clone()
            labels[mask.reshape(-1) == 0] = self.ignore_index
        if self.fp32_upcast:
            logits = logits.float()
        return torch.nn.functional.cross_entropy(
            logits, labels, ignore_index=self.ignore_index, reduction="sum"
        )
