"""BLOOM causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
ALiBi attention (per-head slope × key position, shift-invariant under
softmax), PER-HEAD fused ``query_key_value`` ([H,3,D] rows), biased
LayerNorms incl. a post-embedding ``word_embeddings_layernorm``, tanh-GELU
MLP, tied head. HF keys match BloomForCausalLM (parity-tested).
Attention rides sdpa with an additive alibi+causal bias.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F


def alibi_slopes(num_heads: int) -> torch.Tensor:
    """Per-head ALiBi slopes (power-of-two ladder with interpolated extras)."""
    closest = 2 ** math.floor(math.log2(num_heads))
    base = 2.0 ** (-(2.0 ** -(math.log2(closest) - 3)))
    slopes = torch.pow(torch.tensor(base), torch.arange(1, 1 + closest))
    if closest != num_heads:
        extra_base = 2.0 ** (-(2.0 ** -(math.log2(2 * closest) - 3)))
        n = min(closest, num_heads - closest)
        extra = torch.pow(torch.tensor(extra_base), torch.arange(1, 1 + 2 * n, 2))
        slopes = torch.cat([slopes, extra])
    return slopes.float()


@dataclass
class BloomConfig:
    vocab_size: int = 250880
    hidden_size: int = 64
    n_layer: int = 2
    n_head: int = 8
    layer_norm_epsilon: float = 1e-5
    apply_residual_connection_post_layernorm: bool = False
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.n_head

    @classmethod
    def from_hf_config(cls, hf: Any) -> "BloomConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 250880),
            hidden_size=g("hidden_size", g("n_embed", 64)),
            n_layer=g("n_layer", 2),
            n_head=g("n_head", 8),
            layer_norm_epsilon=g("layer_norm_epsilon", 1e-5),
            apply_residual_connection_post_layernorm=g(
                "apply_residual_connection_post_layernorm", False),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class BloomBlock(nn.Module):
    def __init__(self, cfg: BloomConfig):
        super().__init__()
        E, H, D = cfg.hidden_size, cfg.n_head, cfg.head_dim
        self.H, self.D = H, D
        self.post_ln_residual = cfg.apply_residual_connection_post_layernorm
        self.input_layernorm = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon)
        self.post_attention_layernorm = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon)
        attn = nn.Module()
        attn.query_key_value = nn.Linear(E, 3 * E, bias=True)
        attn.dense = nn.Linear(E, E, bias=True)
        self.self_attention = attn
        mlp = nn.Module()
        mlp.dense_h_to_4h = nn.Linear(E, 4 * E, bias=True)
        mlp.dense_4h_to_h = nn.Linear(4 * E, E, bias=True)
        self.mlp = mlp

    def forward(self, x, bias):
        B, S, E = x.shape
        h = self.input_layernorm(x)
        res = h if self.post_ln_residual else x
        qkv = self.self_attention.query_key_value(h).view(B, S, self.H, 3, self.D)
        q = qkv[..., 0, :].transpose(1, 2)
        k = qkv[..., 1, :].transpose(1, 2)
        v = qkv[..., 2, :].transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, attn_mask=bias)
        x = res + self.self_attention.dense(o.transpose(1, 2).reshape(B, S, E))
        h = self.post_attention_layernorm(x)
        res = h if self.post_ln_residual else x
        return res + self.mlp.dense_4h_to_h(F.gelu(self.mlp.dense_h_to_4h(h),
                                                   approximate="tanh"))


class BloomForCausalLM(nn.Module):
    hf_architectures = ("BloomForCausalLM",)
    config_class = BloomConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> BloomConfig:
        return BloomConfig.from_hf_config(hf_cfg)

    def __init__(self, config: BloomConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = BloomConfig(**config)
        self.config = config
        E = config.hidden_size
        t = nn.Module()
        t.word_embeddings = nn.Embedding(config.vocab_size, E)
        t.word_embeddings_layernorm = nn.LayerNorm(E, eps=config.layer_norm_epsilon)
        t.h = nn.ModuleList(BloomBlock(config) for _ in range(config.n_layer))
        t.ln_f = nn.LayerNorm(E, eps=config.layer_norm_epsilon)
        t.register_buffer("alibi_slopes", alibi_slopes(config.n_head),
                          persistent=False)
        self.transformer = t
        self.lm_head = nn.Linear(E, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = t.word_embeddings.weight
        self.loss_fn = None

    def _bias(self, S: int, device, dtype) -> torch.Tensor:
        # alibi slope_h * key_pos + causal mask, [1, H, S, S]
        j = torch.arange(S, device=device, dtype=torch.float32)
        al = self.transformer.alibi_slopes[:, None, None] * j[None, None, :]
        causal = torch.full((S, S), float("-inf"), device=device)
        causal = torch.triu(causal, diagonal=1)
        return (al + causal[None]).to(dtype)[None]

    def forward(self, input_ids, labels=None, **_: Any):
        t = self.transformer
        B, S = input_ids.shape
        x = t.word_embeddings_layernorm(t.word_embeddings(input_ids))
        bias = self._bias(S, x.device, x.dtype)
        for block in t.h:
            x = block(x, bias)
        hidden = t.ln_f(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            self.transformer.alibi_slopes.copy_(
                alibi_slopes(self.config.n_head).to(
                    self.transformer.alibi_slopes.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, nn.LayerNorm):
                nn.init.ones_(mod.weight)
                nn.init.zeros_(mod.bias)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.transformer.word_embeddings.weight
