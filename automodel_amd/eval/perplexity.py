"""Token-level NLL / perplexity evaluation.

Reference behavior: the reference's validation loop reports val loss per
token (recipes/llm/train_ft.py validation cadence); this utility packages
the same token-sum NLL as a standalone evaluator usable outside a recipe
(rank-shardable like eval/tool_calling.py)."""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F


@torch.no_grad()
def evaluate_nll(model, batches, device=None, max_batches: int | None = None):
    """batches: iterable of dicts with input_ids [B, S] (and optional labels,
    already shifted; defaults to next-token). -> dict with token-mean nll,
    perplexity, token count."""
    was_training = model.training
    model.eval()
    if device is None:
        device = next(model.parameters()).device
    total_nll, total_tok = 0.0, 0
    for i, batch in enumerate(batches):
        if max_batches is not None and i >= max_batches:
            break
        ids = batch["input_ids"].to(device)
        if "labels" in batch:
            inp, labels = ids, batch["labels"].to(device)
        else:
            inp, labels = ids[:, :-1], ids[:, 1:]
        logits = model(inp)
        V = logits.shape[-1]
        nll = F.cross_entropy(logits.reshape(-1, V).float(),
                              labels.reshape(-1), ignore_index=-100,
                              reduction="sum")
        total_nll += float(nll)
        total_tok += int((labels != -100).sum())
    if was_training:
        model.train()
    mean = total_nll / max(total_tok, 1)
    return {"nll_per_token": mean,
            "perplexity": math.exp(min(mean, 50.0)),
            "n_tokens": total_tok}
