"""Multi-token prediction (MTP) loss — DeepSeek-V3-style auxiliary head.

Reference behavior: nemo_automodel/components/loss/mtp.py:48-431
(calculate_mtp_loss over per-depth hidden states) and the custom models'
common/mtp module. One depth here: an extra norm+projection+block combines
h_t with emb(t+1) to predict token t+2; the loss rides the same fused
linear-CE path and is added to the main loss with a coefficient.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from automodel_amd.loss.linear_ce import fused_linear_cross_entropy
from automodel_amd.ops.rms_norm import RMSNorm

IGNORE_INDEX = -100


class MTPHead(nn.Module):
    """h'_t = W_proj [norm(h_t) ; norm(emb(x_{t+1}))] -> one decoder-ish MLP."""

    def __init__(self, hidden_size: int, eps: float = 1e-6):
        super().__init__()
        self.hnorm = RMSNorm(hidden_size, eps, backend="torch")
        self.enorm = RMSNorm(hidden_size, eps, backend="torch")
        self.eh_proj = nn.Linear(2 * hidden_size, hidden_size, bias=False)
        self.final_norm = RMSNorm(hidden_size, eps, backend="torch")

    def forward(self, hidden: torch.Tensor, next_embeds: torch.Tensor) -> torch.Tensor:
        h = torch.cat([self.hnorm(hidden), self.enorm(next_embeds)], dim=-1)
        return self.final_norm(self.eh_proj(h))


def calculate_mtp_loss(
    hidden: torch.Tensor,          # [B, S, H] final hidden states
    embed_tokens: nn.Embedding,
    lm_head_weight: torch.Tensor,
    mtp_head: MTPHead,
    input_ids: torch.Tensor,       # [B, S]
    labels: torch.Tensor,          # [B, S] next-token labels (shifted by 1)
    loss_backend: str = "chunked",
) -> torch.Tensor:
    """Token-sum loss for predicting labels shifted one MORE position.

    position t uses hidden[t] + emb(input_ids[t+1]) to predict labels[t+1].
    """
    B, S, H = hidden.shape
    h = hidden[:, : S - 1]
    nxt = embed_tokens(input_ids[:, 1:])
    mtp_hidden = mtp_head(h, nxt)
    mtp_labels = labels[:, 1:].contiguous()
    return fused_linear_cross_entropy(
        mtp_hidden, lm_head_weight, mtp_labels, backend=loss_backend)
