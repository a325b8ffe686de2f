"""MPT causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
MosaicML ALiBi variant (slope ladder 1/2^(i·8/H₂), odd/even reorder for
non-power-of-2 head counts; bias relative to the LAST key — equivalent
under softmax shift-invariance), full-width fused ``Wqkv`` (chunk 3),
optional ``clip_qkv``, bias-FREE LayerNorms and projections, exact-GELU
4× MLP, tied head. HF keys match MptForCausalLM (parity-tested).
Attention rides sdpa with an additive alibi+causal bias.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F


def mpt_alibi_slopes(num_heads: int, alibi_bias_max: float = 8.0) -> torch.Tensor:
    pow2 = 2 ** math.ceil(math.log2(num_heads))
    base = torch.arange(1, pow2 + 1).float() * (alibi_bias_max / pow2)
    slopes = 1.0 / torch.pow(2, base)
    if pow2 != num_heads:
        slopes = torch.cat([slopes[1::2], slopes[::2]])[:num_heads]
    return slopes


@dataclass
class MptConfig:
    vocab_size: int = 50368
    d_model: int = 2048
    n_layers: int = 24
    n_heads: int = 16
    clip_qkv: float | None = None
    alibi_bias_max: float = 8.0
    layer_norm_epsilon: float = 1e-5
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.d_model // self.n_heads

    @property
    def hidden_size(self):
        return self.d_model

    @property
    def num_hidden_layers(self):
        return self.n_layers

    @classmethod
    def from_hf_config(cls, hf: Any) -> "MptConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        attn_cfg = g("attn_config") or {}
        if hasattr(attn_cfg, "to_dict"):
            attn_cfg = attn_cfg.to_dict()
        return cls(
            vocab_size=g("vocab_size", 50368),
            d_model=g("d_model", 2048),
            n_layers=g("n_layers", 24),
            n_heads=g("n_heads", 16),
            clip_qkv=attn_cfg.get("clip_qkv"),
            alibi_bias_max=attn_cfg.get("alibi_bias_max", 8.0),
            layer_norm_epsilon=g("layer_norm_epsilon", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class MptBlock(nn.Module):
    def __init__(self, cfg: MptConfig):
        super().__init__()
        E, H, D = cfg.d_model, cfg.n_heads, cfg.head_dim
        self.H, self.D = H, D
        self.clip = cfg.clip_qkv
        self.norm_1 = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon, bias=False)
        self.norm_2 = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon, bias=False)
        attn = nn.Module()
        attn.Wqkv = nn.Linear(E, 3 * E, bias=False)
        attn.out_proj = nn.Linear(E, E, bias=False)
        self.attn = attn
        ffn = nn.Module()
        ffn.up_proj = nn.Linear(E, 4 * E, bias=False)
        ffn.down_proj = nn.Linear(4 * E, E, bias=False)
        self.ffn = ffn

    def forward(self, x, bias):
        B, S, E = x.shape
        h = self.norm_1(x)
        qkv = self.attn.Wqkv(h)
        if self.clip:
            qkv = qkv.clamp(-self.clip, self.clip)
        q, k, v = qkv.chunk(3, dim=-1)
        q = q.view(B, S, self.H, self.D).transpose(1, 2)
        k = k.view(B, S, self.H, self.D).transpose(1, 2)
        v = v.view(B, S, self.H, self.D).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, attn_mask=bias)
        x = x + self.attn.out_proj(o.transpose(1, 2).reshape(B, S, E))
        return x + self.ffn.down_proj(F.gelu(self.ffn.up_proj(self.norm_2(x))))


class MptForCausalLM(nn.Module):
    hf_architectures = ("MptForCausalLM",)
    config_class = MptConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> MptConfig:
        return MptConfig.from_hf_config(hf_cfg)

    def __init__(self, config: MptConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = MptConfig(**config)
        self.config = config
        E = config.d_model
        t = nn.Module()
        t.wte = nn.Embedding(config.vocab_size, E)
        t.blocks = nn.ModuleList(MptBlock(config) for _ in range(config.n_layers))
        t.norm_f = nn.LayerNorm(E, eps=config.layer_norm_epsilon, bias=False)
        t.register_buffer("alibi_slopes",
                          mpt_alibi_slopes(config.n_heads, config.alibi_bias_max),
                          persistent=False)
        self.transformer = t
        self.lm_head = nn.Linear(E, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = t.wte.weight
        self.loss_fn = None

    def _bias(self, S: int, device, dtype) -> torch.Tensor:
        # slope_h * (key_pos - S + 1) + causal, [1, H, S, S]
        j = torch.arange(1 - S, 1, device=device, dtype=torch.float32)
        al = self.transformer.alibi_slopes[:, None, None].to(device) * j[None, None, :]
        causal = torch.triu(torch.full((S, S), float("-inf"), device=device),
                            diagonal=1)
        return (al + causal[None]).to(dtype)[None]

    def forward(self, input_ids, labels=None, **_: Any):
        t = self.transformer
        B, S = input_ids.shape
        x = t.wte(input_ids)
        bias = self._bias(S, x.device, x.dtype)
        for block in t.blocks:
            x = block(x, bias)
        hidden = t.norm_f(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            self.transformer.alibi_slopes.copy_(
                mpt_alibi_slopes(self.config.n_heads,
                                 self.config.alibi_bias_max).to(
                    self.transformer.alibi_slopes.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, nn.LayerNorm):
                nn.init.ones_(mod.weight)
                if mod.bias is not None:
                    nn.init.zeros_(mod.bias)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.transformer.wte.weight
