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


class StaticKVCache(KVCache):
    """Static-shape variant for hipGraph decode: attention always runs over
    the full preallocated [max_len] buffers with an additive mask computed
    from a DEVICE position counter — every tensor shape in the decode step
    is constant, so one torch.cuda.graph (hipGraph on ROCm) captures the
    whole per-token forward and replays it with a single hipGraphLaunch
    (guide: capture launch-bound inner loops in hipGraphs).

    Prefill runs with host indexing (not captured); ``freeze_for_graph()``
    moves the position to a device tensor for the captured decode loop —
    the in-graph ``advance_device()`` increments it with no host sync.
    """

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.pos_dev: torch.Tensor | None = None
        self._mask: torch.Tensor | None = None

    @property
    def is_static(self) -> bool:
        return self.pos_dev is not None

    def freeze_for_graph(self) -> None:
        self.pos_dev = torch.tensor(self.pos, device=self.k.device,
                                    dtype=torch.long)
        # persistent additive mask, updated ONCE per step in-graph
        # (building it per layer costs 5 elementwise kernels x n_layers)
        ar = torch.arange(self.max_len, device=self.k.device)
        self._mask = torch.where(ar <= self.pos, 0.0, float("-inf")) \
            .reshape(1, 1, 1, -1).to(self.k.dtype).contiguous()
        self._zero = torch.zeros(1, dtype=self.k.dtype, device=self.k.device)

    def update(self, k_new: torch.Tensor, v_new: torch.Tensor):
        i = self._layer
        self._layer += 1
        if self.pos_dev is None:
            S = k_new.shape[1]
            assert self.pos + S <= self.max_len, "KV cache overflow"
            self.k[i][:, self.pos : self.pos + S] = k_new
            self.v[i][:, self.pos : self.pos + S] = v_new
        else:  # graph mode: single-token write at a device index
            assert k_new.shape[1] == 1, "static decode writes one token"
            idx = self.pos_dev.reshape(1)
            self.k[i].index_copy_(1, idx, k_new)
            self.v[i].index_copy_(1, idx, v_new)
        return self.k[i], self.v[i]          # full static buffers

    def position_ids(self) -> torch.Tensor:
        """[1, 1] device position for rope indexing inside the graph."""
        return self.pos_dev.reshape(1, 1)

    def attn_mask(self) -> torch.Tensor:
        """[1, 1, 1, max_len] additive mask: positions <= pos visible."""
        return self._mask

    def advance_device(self) -> None:
        """In-graph per-step bookkeeping: reveal the next slot, bump pos —
        two tiny kernels per STEP (not per layer)."""
        nxt = torch.clamp(self.pos_dev + 1, max=self.max_len - 1)
        self._mask.view(-1).index_copy_(0, nxt, self._zero)
        self.pos_dev.add_(1)


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
