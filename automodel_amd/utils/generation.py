"""Greedy/sampled decoding for in-loop evaluation and smoke tests.

Reference behavior: the reference's eval path generates with the HF model
(KV-cached ``model.generate``). ``generate`` recomputes the prefix each
step (simple, always correct for any module); ``generate_cached`` runs
prefill-then-decode against the native preallocated KV cache
(utils/kv_cache.py) — O(T) per new token instead of O(T^2).
"""

from __future__ import annotations

import torch


def _pick_next(next_logits: torch.Tensor, temperature: float) -> torch.Tensor:
    if temperature > 0:
        probs = torch.softmax(next_logits / temperature, dim=-1)
        return torch.multinomial(probs, 1)
    return next_logits.argmax(-1, keepdim=True)


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
        nxt = _pick_next(logits[:, -1].float(), temperature)
        ids = torch.cat([ids, nxt], dim=1)
        if eos_token_id is not None and bool((nxt == eos_token_id).all()):
            break
    model.train()
    return ids


@torch.no_grad()
def generate_cached(
    model,
    input_ids: torch.Tensor,
    max_new_tokens: int = 32,
    temperature: float = 0.0,
    eos_token_id: int | None = None,
) -> torch.Tensor:
    """KV-cached decode: one prefill forward over the prompt, then one
    single-token forward per generated token. Greedy output is identical to
    ``generate``. Requires a model whose attention consults the cache
    context (the in-tree model families do)."""
    from automodel_amd.utils.kv_cache import KVCache, kv_cache_context

    model.eval()
    B, T = input_ids.shape
    cache = KVCache.for_model(model, B, T + max_new_tokens)
    ids = input_ids
    with kv_cache_context(cache):
        cache.begin_forward()
        logits = model(input_ids)               # prefill
        cache.advance(T)
        nxt = _pick_next(logits[:, -1].float(), temperature)
        ids = torch.cat([ids, nxt], dim=1)
        for _ in range(max_new_tokens - 1):
            if eos_token_id is not None and bool((nxt == eos_token_id).all()):
                break
            pos = torch.arange(cache.pos, cache.pos + 1,
                               device=ids.device).unsqueeze(0)
            cache.begin_forward()
            logits = model(nxt, position_ids=pos)  # one-token decode
            cache.advance(1)
            nxt = _pick_next(logits[:, -1].float(), temperature)
            ids = torch.cat([ids, nxt], dim=1)
    model.train()
    return ids
