"""Greedy/sampled decoding for in-loop evaluation and smoke tests.

Reference behavior: the reference's eval path generates with the HF model;
here a minimal KV-less decode loop (recompute per step — fine for short eval
generations; a KV-cache decode path is a serving-round feature).
"""

from __future__ import annotations

import torch


@torch.no_grad()
def generate(
    model,
    input_ids: torch.Tensor,
    max_new_tokens: int = 32,
    temperature: float = 0.0,
    eos_token_id: int | None = None,
) -> torch.Tensor:
    model.eval()
    ids = input_ids
    for _ in range(max_new_tokens):
        logits = model(ids)
        next_logits = logits[:, -1].float()
        if temperature > 0:
            probs = torch.softmax(next_logits / temperature, dim=-1)
            nxt = torch.multinomial(probs, 1)
        else:
            nxt = next_logits.argmax(-1, keepdim=True)
        ids = torch.cat([ids, nxt], dim=1)
        if eos_token_id is not None and bool((nxt == eos_token_id).all()):
            break
    model.train()
    return ids
