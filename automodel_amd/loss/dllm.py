"""Discrete-diffusion LM (MDLM-style) masking + loss.

Reference behavior: nemo_automodel/components/loss/dllm_loss.py:104
(MDLMCrossEntropyLoss: CE on corrupted-and-supervised positions weighted by
the scheduler weight 1/t, normalized by total supervised tokens) and the
dllm recipe's corruption step (sample t ~ U(eps, 1) per sequence, replace
tokens with the mask id with probability t; the model denoises
bidirectionally). Independent implementation with the same math.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


def mdlm_corrupt(input_ids: torch.Tensor, mask_token_id: int,
                 loss_mask: torch.Tensor | None = None, eps: float = 1e-3,
                 generator: torch.Generator | None = None):
    """-> (noisy_ids, noise_mask [B,L] bool, p_mask [B,L]). One t per
    sequence, linear schedule; only supervised positions are corruptible."""
    B, L = input_ids.shape
    t = torch.rand(B, 1, device=input_ids.device, generator=generator) \
        .clamp_min(eps)
    p_mask = t.expand(B, L)
    corrupt = torch.rand(B, L, device=input_ids.device, generator=generator) < p_mask
    if loss_mask is not None:
        corrupt &= loss_mask.bool()
    noisy = torch.where(corrupt, torch.full_like(input_ids, mask_token_id),
                        input_ids)
    return noisy, corrupt, p_mask


class MDLMCrossEntropyLoss(nn.Module):
    """loss = sum_{i in masked & supervised} CE_i / t  /  n_supervised."""

    def forward(self, logits: torch.Tensor, target_ids: torch.Tensor,
                noise_mask: torch.Tensor, p_mask: torch.Tensor,
                loss_mask: torch.Tensor | None = None,
                num_diffusion_tokens: int | None = None) -> torch.Tensor:
        V = logits.shape[-1]
        nll = F.cross_entropy(logits.reshape(-1, V).float(),
                              target_ids.reshape(-1), reduction="none") \
            .reshape_as(target_ids)
        mask = noise_mask
        if loss_mask is not None:
            mask = mask & loss_mask.bool()
        weighted = nll * mask.float() / p_mask.clamp_min(1e-8)
        loss = weighted.sum()
        denom = num_diffusion_tokens
        if denom is None:
            denom = int(mask.sum()) if loss_mask is None else int(loss_mask.sum())
        return loss / max(denom, 1)
