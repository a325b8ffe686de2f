"""Preallocated KV cache for incremental decode.

Reference behavior: the reference's generation-based eval decodes through HF
``model.generate`` (KV-cached); this framework's models are custom modules,
so the cache is native: one preallocated [L, B, max_len, Hkv, D] pair sized
for 288 GB HBM3E (a Llama-3-8B cache at 8k ctx, B=8 is ~4 GB in bf16 —
preallocate once, no per-step allocation or copies beyond the new slots).

Used via a context so module signatures stay unchanged (same pattern as the
varlen context in ops/attention.py): attention layers call
``maybe_update_kv(k, v)`` after rope; under an active cache it returns the
full prefix k/v and records the new slots; otherwise it is a no-op.
"""

from __future__ import annotations

from contextlib import contextmanager

import torch

_ACTIVE: "KVCache | None" = None


class KVCache:
    def __init__(self, n_layers: int, batch: int, max_len: int,
                 n_kv_heads: int, head_dim: int, device, dtype):
        shape = (n_layers, batch, max_len, n_kv_heads, head_dim)
        self.k = torch.zeros(shape, device=device, dtype=dtype)
        self.v = torch.zeros(shape, device=device, dtype=dtype)
        self.max_len = max_len
        self.pos = 0
        self._layer = 0

    @classmethod
    def for_model(cls, model, batch: int, max_len: int) -> "KVCache":
        cfg = model.config
        p = next(model.parameters())
        return cls(cfg.num_hidden_layers, batch, max_len,
                   cfg.num_key_value_heads, cfg.head_dim, p.device, p.dtype)

    def begin_forward(self) -> None:
        """Call before each model forward; layers consume slots in order."""
        self._layer = 0

    def update(self, k_new: torch.Tensor, v_new: torch.Tensor):
        """Record [B,S,Hk,D] at the current position; -> (k, v) over
        positions [0, pos+S) for attention."""
        i = self._layer
        self._layer += 1
        S = k_new.shape[1]
        assert self.pos + S <= self.max_len, "KV cache overflow"
        self.k[i][:, self.pos : self.pos + S] = k_new
        self.v[i][:, self.pos : self.pos + S] = v_new
        return (self.k[i][:, : self.pos + S], self.v[i][:, : self.pos + S])

    def advance(self, n: int) -> None:
        self.pos += n


@contextmanager
def kv_cache_context(cache: KVCache):
    global _ACTIVE
    prev = _ACTIVE
    _ACTIVE = cache
    try:
        yield cache
    finally:
        _ACTIVE = prev


def active_kv_cache() -> KVCache | None:
    return _ACTIVE


def maybe_update_kv(k: torch.Tensor, v: torch.Tensor):
    """Attention-layer hook: (k, v, cache_pos). cache_pos is the q_start
    offset for causal masking (0 when no cache is active)."""
    if _ACTIVE is None:
        return k, v, 0
    pos = _ACTIVE.pos
    k_full, v_full = _ACTIVE.update(k, v)
    return k_full, v_full, pos
