"""Falcon causal LM (new decoder architecture), MI355X-native.

Reference behavior: covered by the reference's HF-wrapped model path; here
native. Falcon-40B/180B-style blocks: PARALLEL attention + MLP residual
(x + attn(ln_attn(x)) + mlp(ln_mlp(x))), fused query_key_value whose rows
interleave per KV group ([q x (H/kv), k, v] blocks), rotary embeddings,
GELU MLP, biased LayerNorms, no biases on the projections.

HF keys match FalconForCausalLM (new_decoder_architecture=True,
parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache


@dataclass
class FalconConfig:
    vocab_size: int = 65024
    hidden_size: int = 8192
    num_hidden_layers: int = 60
    num_attention_heads: int = 128
    num_kv_heads: int = 8
    max_position_embeddings: int = 2048
    rope_theta: float = 10000.0
    layer_norm_epsilon: float = 1e-5
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @property
    def num_key_value_heads(self):
        return self.num_kv_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "FalconConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 65024),
            hidden_size=g("hidden_size", 8192),
            num_hidden_layers=g("num_hidden_layers", 60),
            num_attention_heads=g("num_attention_heads", 128),
            num_kv_heads=g("num_kv_heads", 8),
            max_position_embeddings=g("max_position_embeddings", 2048),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            layer_norm_epsilon=g("layer_norm_epsilon", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class FalconBlock(nn.Module):
    def __init__(self, cfg: FalconConfig):
        super().__init__()
        E, H, Hk, D = (cfg.hidden_size, cfg.num_attention_heads,
                       cfg.num_kv_heads, cfg.head_dim)
        self.H, self.Hk, self.D = H, Hk, D
        self.ln_attn = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon)
        self.ln_mlp = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon)
        attn = nn.Module()
        attn.query_key_value = nn.Linear(E, (H + 2 * Hk) * D, bias=False)
        attn.dense = nn.Linear(H * D, E, bias=False)
        self.self_attention = attn
        mlp = nn.Module()
        mlp.dense_h_to_4h = nn.Linear(E, 4 * E, bias=False)
        mlp.dense_4h_to_h = nn.Linear(4 * E, E, bias=False)
        self.mlp = mlp

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        h = self.ln_attn(x)
        # fused qkv interleaved per kv group: [q x (H/Hk), k, v]
        per = self.H // self.Hk
        qkv = self.self_attention.query_key_value(h) \
            .view(B, S, self.Hk, per + 2, self.D)
        q = qkv[..., :per, :].reshape(B, S, self.H, self.D)
        k = qkv[..., per, :]          # [B, S, Hk, D]
        v = qkv[..., per + 1, :]
        q, k = apply_rope_ref(q, k, cos, sin)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        attn_out = self.self_attention.dense(
            o.transpose(1, 2).reshape(B, S, -1))
        mlp_out = self.mlp.dense_4h_to_h(
            F.gelu(self.mlp.dense_h_to_4h(self.ln_mlp(x))))
        return x + attn_out + mlp_out          # parallel residual


class FalconForCausalLM(nn.Module):
    hf_architectures = ("FalconForCausalLM",)
    config_class = FalconConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> FalconConfig:
        return FalconConfig.from_hf_config(hf_cfg)

    def __init__(self, config: FalconConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = FalconConfig(**config)
        self.config = config
        t = nn.Module()
        t.word_embeddings = nn.Embedding(config.vocab_size, config.hidden_size)
        t.h = nn.ModuleList(FalconBlock(config)
                            for _ in range(config.num_hidden_layers))
        t.ln_f = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)
        self.transformer = t
        cos, sin = build_rope_cache(config.head_dim,
                                    config.max_position_embeddings,
                                    config.rope_theta)
        t.register_buffer("rope_cos", cos, persistent=False)
        t.register_buffer("rope_sin", sin, persistent=False)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = t.word_embeddings.weight
        self.loss_fn = None

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        t = self.transformer
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = t.rope_cos[:S], t.rope_sin[:S]
        else:
            cos, sin = t.rope_cos[position_ids[0]], t.rope_sin[position_ids[0]]
        cos, sin = cos.float(), sin.float()
        x = t.word_embeddings(input_ids)
        for block in t.h:
            x = block(x, cos, sin)
        hidden = t.ln_f(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            cos, sin = build_rope_cache(self.config.head_dim,
                                        self.config.max_position_embeddings,
                                        self.config.rope_theta)
            self.transformer.rope_cos.copy_(cos.to(self.transformer.rope_cos.device))
            self.transformer.rope_sin.copy_(sin.to(self.transformer.rope_sin.device))
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
