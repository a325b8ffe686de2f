"""Arcee (AFM) causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Llama-shaped pre-norm RMS blocks with a GATELESS relu² MLP
(down(relu(up(x))²)) and standard half-split rotary. HF keys match
ArceeForCausalLM (parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache


@dataclass
class ArceeConfig:
    vocab_size: int = 32000
    hidden_size: int = 2560
    intermediate_size: int = 18432
    num_hidden_layers: int = 36
    num_attention_heads: int = 20
    num_key_value_heads: int = 20
    head_dim: int = 128
    max_position_embeddings: int = 4096
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-5
    attention_bias: bool = False
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "ArceeConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 32000),
            hidden_size=g("hidden_size", 2560),
            intermediate_size=g("intermediate_size", 18432),
            num_hidden_layers=g("num_hidden_layers", 36),
            num_attention_heads=g("num_attention_heads", 20),
            num_key_value_heads=g("num_key_value_heads") or g("num_attention_heads", 20),
            head_dim=g("head_dim") or g("hidden_size", 2560) // g("num_attention_heads", 20),
            max_position_embeddings=g("max_position_embeddings", 4096),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            attention_bias=g("attention_bias", False),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class ArceeLayer(nn.Module):
    def __init__(self, cfg: ArceeConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        E = cfg.hidden_size
        self.input_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        b = cfg.attention_bias
        attn = nn.Module()
        attn.q_proj = nn.Linear(E, H * D, bias=b)
        attn.k_proj = nn.Linear(E, Hk * D, bias=b)
        attn.v_proj = nn.Linear(E, Hk * D, bias=b)
        attn.o_proj = nn.Linear(H * D, E, bias=b)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.up_proj = nn.Linear(E, cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, E, bias=False)
        self.mlp = mlp

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        h = self.input_layernorm(x)
        a = self.self_attn
        q = a.q_proj(h).view(B, S, self.H, self.D)
        k = a.k_proj(h).view(B, S, self.Hk, self.D)
        v = a.v_proj(h).view(B, S, self.Hk, self.D)
        q, k = apply_rope_ref(q, k, cos, sin)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        x = x + a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        h = self.post_attention_layernorm(x)
        return x + self.mlp.down_proj(F.relu(self.mlp.up_proj(h)).square())


class ArceeForCausalLM(nn.Module):
    hf_architectures = ("ArceeForCausalLM",)
    config_class = ArceeConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> ArceeConfig:
        return ArceeConfig.from_hf_config(hf_cfg)

    def __init__(self, config: ArceeConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = ArceeConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(ArceeLayer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.norm = RMSNorm(config.hidden_size, eps=config.rms_norm_eps)
        cos, sin = build_rope_cache(config.head_dim,
                                    config.max_position_embeddings,
                                    config.rope_theta)
        inner.register_buffer("rope_cos", cos, persistent=False)
        inner.register_buffer("rope_sin", sin, persistent=False)
        self.model = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = inner.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None, **_: Any):
        m = self.model
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = m.rope_cos[:S], m.rope_sin[:S]
        else:
            cos, sin = m.rope_cos[position_ids[0]], m.rope_sin[position_ids[0]]
        cos, sin = cos.float(), sin.float()
        x = m.embed_tokens(input_ids)
        for layer in m.layers:
            x = layer(x, cos, sin)
        hidden = m.norm(x)
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
            self.model.rope_cos.copy_(cos.to(self.model.rope_cos.device))
            self.model.rope_sin.copy_(sin.to(self.model.rope_sin.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
