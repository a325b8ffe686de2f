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
def generate_graphed(
    model,
    input_ids: torch.Tensor,
    max_new_tokens: int = 32,
    use_hip_graph: bool | None = None,
) -> torch.Tensor:
    """Greedy decode with a STATIC KV cache: every per-token forward has
    constant shapes, so on GPU the whole step (embed → 32 layers → argmax →
    feedback into the input buffer → position increment) is captured in ONE
    hipGraph and each token costs a single hipGraphLaunch instead of
    hundreds of kernel launches. CPU falls back to eager static-cache math
    (same numerics — tested for parity with ``generate``)."""
    from automodel_amd.utils.kv_cache import StaticKVCache, kv_cache_context

    model.eval()
    B, T = input_ids.shape
    dev = input_ids.device
    use_hip_graph = (dev.type == "cuda") if use_hip_graph is None else use_hip_graph
    cache = StaticKVCache(model.config.num_hidden_layers, B, T + max_new_tokens,
                          model.config.num_key_value_heads, model.config.head_dim,
                          dev, next(model.parameters()).dtype)
    with kv_cache_context(cache):
        cache.begin_forward()
        logits = model(input_ids)                      # prefill (host-indexed)
        assert cache._layer == model.config.num_hidden_layers, (
            "model did not consult the KV cache — static decode unsupported")
        cache.advance(T)
        cache.freeze_for_graph()                       # device position from here
        nxt = logits[:, -1].argmax(-1, keepdim=True)   # [B, 1]
        ids_buf = nxt.clone()                          # static input buffer
        out = torch.empty(B, max_new_tokens, dtype=input_ids.dtype, device=dev)
        out[:, 0] = nxt[:, 0]

        def decode_step():
            cache.begin_forward()
            lg = model(ids_buf, position_ids=cache.position_ids())
            ids_buf.copy_(lg[:, -1].argmax(-1, keepdim=True))
            cache.advance_device()

        if use_hip_graph and max_new_tokens >= 3:
            # warmup on a side stream (allocator state), then capture
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                decode_step()
            torch.cuda.current_stream().wait_stream(s)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                decode_step()
            out[:, 1] = ids_buf[:, 0]  # warmup emitted token 1; capture token 2
            start = 2
            for i in range(start, max_new_tokens):
                g.replay()
                out[:, i] = ids_buf[:, 0]
        else:
            start = 1
            for i in range(start, max_new_tokens):
                decode_step()
                out[:, i] = ids_buf[:, 0]
    model.train()
    return torch.cat([input_ids, out], dim=1)


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
        assert cache._layer == model.config.num_hidden_layers, (
            "model did not consult the KV cache (attention lacks "
            "maybe_update_kv) — cached decode would be silently wrong")
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
