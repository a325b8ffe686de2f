"""DeepSeek-V3.2 (DSA — DeepSeek Sparse Attention), MI355X-native.

Reference behavior: nemo_automodel's sparse-attention model families
(components/models/deepseek_v32/ and glm_moe_dsa/ — lightning indexer +
top-k sparse attention over the MLA stack; SURVEY.md §2.9 row 19
"TileLang sparse"). Architecture per the public DeepSeek-V3.2 design:

  * the full DeepSeek-V3 MLA + MoE stack is reused unchanged;
  * every attention layer gains a LIGHTNING INDEXER: ``H_I`` small query
    heads ``wq``, one shared key ``wk`` per token, and per-query head
    weights ``weights_proj``; score I[t,s] = sum_j w[t,j] relu(qI . kI);
    rope is applied to the leading ``qk_rope_head_dim`` index channels;
  * each query attends only to its ``index_topk`` best-scoring causal
    keys (the dense path runs whenever S <= index_topk — identical
    output, no selection overhead);
  * optional training-time distillation: set ``collect_indexer_loss`` and
    read ``indexer_kl`` per layer — KL(head-summed main attention ||
    indexer distribution), the DSA warmup signal.

No public HF implementation exists in this image's transformers; numerics
are covered by construction tests (topk >= S reproduces the dense
DeepSeek-V3 output bit-for-bit on shared weights).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.deepseek_v3.model import (
    DeepseekV3Config,
    DeepseekV3ForCausalLM,
    MLAAttention,
)
from automodel_amd.ops.sparse_attention import (
    indexer_kl_loss,
    lightning_index_scores,
    sparse_gather_attention,
    topk_causal_indices,
)


@dataclass
class DeepseekV32Config(DeepseekV3Config):
    index_n_heads: int = 64
    index_head_dim: int = 128
    index_topk: int = 2048

    @classmethod
    def from_hf_config(cls, hf: Any) -> "DeepseekV32Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        base = DeepseekV3Config.from_hf_config(hf)
        cfg = cls(**{f: getattr(base, f) for f in base.__dataclass_fields__})
        cfg.index_n_heads = hf.get("index_n_heads", 64)
        cfg.index_head_dim = hf.get("index_head_dim", 128)
        cfg.index_topk = hf.get("index_topk", 2048)
        return cfg


class LightningIndexer(nn.Module):
    """Cheap (query, key) scorer: H_I small query heads, one shared key."""

    def __init__(self, cfg: DeepseekV32Config):
        super().__init__()
        self.n_heads = cfg.index_n_heads
        self.head_dim = cfg.index_head_dim
        self.rope_dim = min(cfg.qk_rope_head_dim, cfg.index_head_dim)
        self.wq = nn.Linear(cfg.hidden_size, cfg.index_n_heads * cfg.index_head_dim,
                            bias=False)
        self.wk = nn.Linear(cfg.hidden_size, cfg.index_head_dim, bias=False)
        self.k_norm = nn.LayerNorm(cfg.index_head_dim)
        self.weights_proj = nn.Linear(cfg.hidden_size, cfg.index_n_heads,
                                      bias=False)

    @staticmethod
    def _rope(t, cos, sin, r):
        rot, keep = t[..., :r], t[..., r:]
        t1, t2 = rot.chunk(2, dim=-1)
        rh = torch.cat([-t2, t1], dim=-1)
        c = cos[..., :r].unsqueeze(-2) if cos.dim() == 2 else cos[:, :, None, :r]
        s = sin[..., :r].unsqueeze(-2) if sin.dim() == 2 else sin[:, :, None, :r]
        return torch.cat([rot * c + rh * s, keep], dim=-1)

    def forward(self, x: torch.Tensor, cos: torch.Tensor,
                sin: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        q = self.wq(x).view(B, S, self.n_heads, self.head_dim)
        k = self.k_norm(self.wk(x))                       # [B, S, D_I]
        r = self.rope_dim
        cf, sf = cos.float(), sin.float()
        q = self._rope(q.float(), cf, sf, r)
        k = self._rope(k.float().unsqueeze(2), cf, sf, r).squeeze(2)
        w = self.weights_proj(x)                          # [B, S, H_I]
        return lightning_index_scores(q, k, w)


class DSAttention(MLAAttention):
    """MLA attention restricted to the indexer's top-k causal keys."""

    def __init__(self, cfg: DeepseekV32Config, backend: BackendConfig):
        super().__init__(cfg, backend)
        self.indexer = LightningIndexer(cfg)
        self.index_topk = cfg.index_topk
        self.collect_indexer_loss = False
        self.indexer_kl: torch.Tensor | None = None

    def forward(self, x: torch.Tensor, cos: torch.Tensor,
                sin: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        B, S, _ = x.shape
        H = self.num_heads
        qf, kf, v = self._qkv(x, cos, sin)
        scores = self.indexer(x, cos, sin)
        if self.collect_indexer_loss:
            att = torch.einsum("bshd,bkhd->bshk", qf.float(), kf.float())
            att = att * self.scale
            causal = torch.ones(S, S, dtype=torch.bool, device=x.device).tril()
            att = att.masked_fill(~causal[:, None], float("-inf"))
            probs = att.softmax(-1).sum(2)               # head-summed [B,S,S]
            self.indexer_kl = indexer_kl_loss(scores, probs)
        if S <= self.index_topk:
            o = torch.nn.functional.scaled_dot_product_attention(
                qf.transpose(1, 2), kf.transpose(1, 2), v.transpose(1, 2),
                is_causal=True, scale=self.scale)
            o = o.transpose(1, 2)
        else:
            idx, valid = topk_causal_indices(scores.detach(), self.index_topk)
            o = sparse_gather_attention(qf, kf, v.contiguous(), idx, valid,
                                        self.scale)
        return self.o_proj(o.reshape(B, S, H * cfg.v_head_dim))


class DeepseekV32ForCausalLM(DeepseekV3ForCausalLM):
    hf_architectures = ("DeepseekV32ForCausalLM",)
    config_class = DeepseekV32Config

    @staticmethod
    def config_from_hf(hf_cfg) -> DeepseekV32Config:
        return DeepseekV32Config.from_hf_config(hf_cfg)

    def __init__(self, config: DeepseekV32Config | dict,
                 backend: BackendConfig | dict | None = None):
        if isinstance(config, dict):
            config = DeepseekV32Config(**config)
        super().__init__(config, backend)
        bk = self.backend
        for layer in self.model.layers:
            layer.self_attn = DSAttention(config, bk)

    def set_collect_indexer_loss(self, flag: bool) -> None:
        for layer in self.model.layers:
            layer.self_attn.collect_indexer_loss = flag

    def indexer_losses(self) -> list[torch.Tensor]:
        return [layer.self_attn.indexer_kl for layer in self.model.layers
                if layer.self_attn.indexer_kl is not None]
