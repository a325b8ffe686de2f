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


def _roll_mask(t: torch.Tensor, k: int, fill) -> torch.Tensor:
    """roll(t, -k) along seq with the trailing k positions set to ``fill``."""
    out = torch.roll(t, shifts=-k, dims=1).clone()
    out[:, -k:] = fill
    return out


class MTPHeads(nn.Module):
    """Chained multi-depth MTP heads (DeepSeek-V3 section 2.2: depth k
    consumes depth k-1's hidden plus the (t+k)-shifted token embedding).
    Reference: components/models/common/mtp/mtp.py depth iteration."""

    def __init__(self, hidden_size: int, n_depths: int = 1, eps: float = 1e-6):
        super().__init__()
        self.heads = nn.ModuleList(MTPHead(hidden_size, eps)
                                   for _ in range(n_depths))

    def __len__(self):
        return len(self.heads)


def calculate_mtp_loss_multi(
    hidden: torch.Tensor,          # [B, S, H]
    embed_tokens: nn.Embedding,
    lm_head_weight: torch.Tensor,
    mtp_heads: MTPHeads,
    input_ids: torch.Tensor,       # [B, S]
    labels: torch.Tensor,          # [B, S] next-token labels
    scaling_factor: float = 0.1,
    seq_idx: torch.Tensor | None = None,
    cu_seqlens: torch.Tensor | None = None,
    loss_backend: str = "chunked",
    return_per_depth: bool = False,
):
    """Multi-depth MTP loss with packed-sequence boundary masking.

    Depth k (1-based) predicts ``labels`` rolled k positions left; a rolled
    position is IGNORED when its source lies beyond the sequence tail or in
    a different packed sub-sequence (reference loss/mtp.py:107 — seq_idx /
    cu_seqlens guard). Returns ``scaling_factor * sum_k CE_k`` (token-sum),
    optionally with the unscaled per-depth losses.
    """
    B, S, H = hidden.shape
    if seq_idx is None and cu_seqlens is not None:
        pos = torch.arange(S, device=hidden.device)
        seq_idx = torch.bucketize(pos, cu_seqlens[1:-1].to(hidden.device),
                                  right=True).unsqueeze(0).expand(B, -1)
    cur = hidden
    per_depth = []
    total = None
    for k, head in enumerate(mtp_heads.heads, start=1):
        ids_k = _roll_mask(input_ids, 1, 0) if k == 1 else _roll_mask(ids_k, 1, 0)
        labels_k = _roll_mask(labels, k, IGNORE_INDEX)
        if seq_idx is not None:
            src_seq = _roll_mask(seq_idx, k, -1)
            labels_k = torch.where(src_seq == seq_idx, labels_k,
                                   torch.full_like(labels_k, IGNORE_INDEX))
        cur = mtp_heads.heads[k - 1](cur, embed_tokens(ids_k))
        lk = fused_linear_cross_entropy(cur, lm_head_weight,
                                        labels_k.contiguous(),
                                        backend=loss_backend)
        per_depth.append(lk)
        total = lk if total is None else total + lk
    total = total * scaling_factor
    return (total, per_depth) if return_per_depth else total
