"""Persimmon (Fuyu LM trunk) causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Biased-LayerNorm pre-norm blocks with PER-HEAD fused ``query_key_value``
([H,3,D] rows), per-head biased qk LayerNorms applied BEFORE rope,
PARTIAL rotary (factor 0.5, half-split), relu² MLP
(dense_h_to_4h/dense_4h_to_h, biased). HF keys match
PersimmonForCausalLM (parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import build_rope_cache


@dataclass
class PersimmonConfig:
    vocab_size: int = 262144
    hidden_size: int = 4096
    intermediate_size: int = 16384
    num_hidden_layers: int = 36
    num_attention_heads: int = 64
    partial_rotary_factor: float = 0.5
    qk_layernorm: bool = True
    max_position_embeddings: int = 16384
    rope_theta: float = 25000.0
    layer_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "PersimmonConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 262144),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 16384),
            num_hidden_layers=g("num_hidden_layers", 36),
            num_attention_heads=g("num_attention_heads", 64),
            partial_rotary_factor=rp.get("partial_rotary_factor",
                                         g("partial_rotary_factor", 0.5)),
            qk_layernorm=g("qk_layernorm", True),
            max_position_embeddings=g("max_position_embeddings", 16384),
            rope_theta=rp.get("rope_theta", g("rope_theta", 25000.0)),
            layer_norm_eps=g("layer_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class PersimmonLayer(nn.Module):
    def __init__(self, cfg: PersimmonConfig):
        super().__init__()
        H, D, E = cfg.num_attention_heads, cfg.head_dim, cfg.hidden_size
        self.H, self.D = H, D
        self.rot = int(D * cfg.partial_rotary_factor)
        self.qk_ln = cfg.qk_layernorm
        self.input_layernorm = nn.LayerNorm(E, eps=cfg.layer_norm_eps)
        self.post_attention_layernorm = nn.LayerNorm(E, eps=cfg.layer_norm_eps)
        attn = nn.Module()
        attn.query_key_value = nn.Linear(E, 3 * H * D, bias=True)
        attn.dense = nn.Linear(H * D, E, bias=True)
        if self.qk_ln:
            attn.q_layernorm = nn.LayerNorm(D, eps=cfg.layer_norm_eps)
            attn.k_layernorm = nn.LayerNorm(D, eps=cfg.layer_norm_eps)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.dense_h_to_4h = nn.Linear(E, cfg.intermediate_size, bias=True)
        mlp.dense_4h_to_h = nn.Linear(cfg.intermediate_size, E, bias=True)
        self.mlp = mlp

    @staticmethod
    def _rot_half(t, cos, sin):
        t1, t2 = t.chunk(2, dim=-1)
        rh = torch.cat([-t2, t1], dim=-1)
        return t * cos + rh * sin

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        h = self.input_layernorm(x)
        a = self.self_attn
        qkv = a.query_key_value(h).view(B, S, self.H, 3, self.D)
        q, k, v = qkv[..., 0, :], qkv[..., 1, :], qkv[..., 2, :]
        if self.qk_ln:
            q = a.q_layernorm(q)
            k = a.k_layernorm(k)
        r = self.rot
        q = torch.cat([self._rot_half(q[..., :r], cos, sin), q[..., r:]], dim=-1)
        k = torch.cat([self._rot_half(k[..., :r], cos, sin), k[..., r:]], dim=-1)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True)
        x = x + a.dense(o.transpose(1, 2).reshape(B, S, -1))
        h = self.post_attention_layernorm(x)
        return x + self.mlp.dense_4h_to_h(
            F.relu(self.mlp.dense_h_to_4h(h)).square())


class PersimmonForCausalLM(nn.Module):
    hf_architectures = ("PersimmonForCausalLM",)
    config_class = PersimmonConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> PersimmonConfig:
        return PersimmonConfig.from_hf_config(hf_cfg)

    def __init__(self, config: PersimmonConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = PersimmonConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(PersimmonLayer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.final_layernorm = nn.LayerNorm(config.hidden_size,
                                             eps=config.layer_norm_eps)
        rot = int(config.head_dim * config.partial_rotary_factor)
        cos, sin = build_rope_cache(rot, config.max_position_embeddings,
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
        cos = cos.float()[None, :, None, :]
        sin = sin.float()[None, :, None, :]
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
