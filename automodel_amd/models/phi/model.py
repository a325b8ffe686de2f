"""Phi-1/2 causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Phi-2-style blocks: PARALLEL attention + MLP over ONE shared input
LayerNorm (x + attn(ln(x)) + mlp(ln(x))), biased q/k/v and a ``dense``
output projection, PARTIAL rotary (factor 0.5 on the leading head dims),
tanh-GELU MLP (fc1/fc2, biased), biased final LayerNorm and lm_head bias.

HF keys match PhiForCausalLM (parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import build_rope_cache


@dataclass
class PhiConfig:
    vocab_size: int = 51200
    hidden_size: int = 2560
    intermediate_size: int = 10240
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int | None = None
    partial_rotary_factor: float = 0.5
    max_position_embeddings: int = 2048
    rope_theta: float = 10000.0
    layer_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.num_key_value_heads is None:
            self.num_key_value_heads = self.num_attention_heads

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "PhiConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 51200),
            hidden_size=g("hidden_size", 2560),
            intermediate_size=g("intermediate_size", 10240),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads"),
            partial_rotary_factor=g("partial_rotary_factor", 0.5),
            max_position_embeddings=g("max_position_embeddings", 2048),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            layer_norm_eps=g("layer_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class PhiDecoderLayer(nn.Module):
    def __init__(self, cfg: PhiConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        self.rot = int(D * cfg.partial_rotary_factor)
        self.input_layernorm = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        attn = nn.Module()
        attn.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=True)
        attn.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=True)
        attn.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=True)
        attn.dense = nn.Linear(H * D, cfg.hidden_size, bias=True)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.fc1 = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=True)
        mlp.fc2 = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=True)
        self.mlp = mlp

    @staticmethod
    def _rot_half(t, cos, sin):
        t1, t2 = t.chunk(2, dim=-1)
        rh = torch.cat([-t2, t1], dim=-1)
        return t * cos + rh * sin

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        h = self.input_layernorm(x)
        a = self.self_attn
        q = a.q_proj(h).view(B, S, self.H, self.D)
        k = a.k_proj(h).view(B, S, self.Hk, self.D)
        v = a.v_proj(h).view(B, S, self.Hk, self.D)
        r = self.rot
        c = cos[None, :, None, :]
        s = sin[None, :, None, :]
        q = torch.cat([self._rot_half(q[..., :r], c, s), q[..., r:]], dim=-1)
        k = torch.cat([self._rot_half(k[..., :r], c, s), k[..., r:]], dim=-1)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        attn_out = a.dense(o.transpose(1, 2).reshape(B, S, -1))
        mlp_out = self.mlp.fc2(F.gelu(self.mlp.fc1(h), approximate="tanh"))
        return x + attn_out + mlp_out           # parallel residual


class PhiForCausalLM(nn.Module):
    hf_architectures = ("PhiForCausalLM",)
    config_class = PhiConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> PhiConfig:
        return PhiConfig.from_hf_config(hf_cfg)

    def __init__(self, config: PhiConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = PhiConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(PhiDecoderLayer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.final_layernorm = nn.LayerNorm(config.hidden_size,
                                             eps=config.layer_norm_eps)
        rot = int(config.head_dim * config.partial_rotary_factor)
        cos, sin = build_rope_cache(rot, config.max_position_embeddings,
                                    config.rope_theta)
        inner.register_buffer("rope_cos", cos, persistent=False)
        inner.register_buffer("rope_sin", sin, persistent=False)
        self.model = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=True)
        if config.tie_word_embeddings:
            self.lm_head.weight = inner.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
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
        hidden = m.final_layernorm(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            rot = int(self.config.head_dim * self.config.partial_rotary_factor)
            cos, sin = build_rope_cache(rot, self.config.max_position_embeddings,
                                        self.config.rope_theta)
            self.model.rope_cos.copy_(cos.to(self.model.rope_cos.device))
            self.model.rope_sin.copy_(sin.to(self.model.rope_sin.device))
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
            self.lm_head.weight = self.model.embed_tokens.weight
