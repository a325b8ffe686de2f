"""Cohere (Command-R) causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Command-R blocks: PARALLEL attention + SwiGLU MLP over ONE shared
bias-free LayerNorm (x + attn(ln(x)) + mlp(ln(x))), INTERLEAVED-pair
rotary (handled with the shared-permutation trick — see
models/llama rope_interleaved), logits scaled by ``logit_scale``, tied
embeddings. HF keys match CohereForCausalLM (parity-tested)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache


@dataclass
class CohereConfig:
    vocab_size: int = 256000
    hidden_size: int = 8192
    intermediate_size: int = 22528
    num_hidden_layers: int = 40
    num_attention_heads: int = 64
    num_key_value_heads: int = 64
    max_position_embeddings: int = 8192
    rope_theta: float = 10000.0
    layer_norm_eps: float = 1e-5
    logit_scale: float = 0.0625
    sliding_window: int | None = None          # Cohere2 (Command-R7B)
    layer_types: tuple = ()                    # "sliding_attention"/"full_attention"
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "CohereConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 256000),
            hidden_size=g("hidden_size", 8192),
            intermediate_size=g("intermediate_size", 22528),
            num_hidden_layers=g("num_hidden_layers", 40),
            num_attention_heads=g("num_attention_heads", 64),
            num_key_value_heads=g("num_key_value_heads", 64),
            max_position_embeddings=g("max_position_embeddings", 8192),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            layer_norm_eps=g("layer_norm_eps", 1e-5),
            logit_scale=g("logit_scale", 0.0625),
            sliding_window=g("sliding_window"),
            layer_types=tuple(g("layer_types") or ()),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class CohereLayerNorm(nn.Module):
    """Mean-variance LayerNorm with weight only (no bias)."""

    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        xf = x.float()
        mu = xf.mean(-1, keepdim=True)
        var = (xf - mu).pow(2).mean(-1, keepdim=True)
        return ((xf - mu) * torch.rsqrt(var + self.eps)).to(x.dtype) * self.weight


class CohereDecoderLayer(nn.Module):
    def __init__(self, cfg: CohereConfig, layer_idx: int = 0):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        # Cohere2: rope ONLY on sliding_attention layers (full layers NoPE)
        kind = cfg.layer_types[layer_idx] if cfg.layer_types else "full_attention"
        self.sliding = cfg.sliding_window if kind == "sliding_attention" else None
        self.use_rope = not cfg.layer_types or kind == "sliding_attention"
        self.input_layernorm = CohereLayerNorm(cfg.hidden_size, cfg.layer_norm_eps)
        attn = nn.Module()
        attn.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=False)
        attn.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        attn.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        attn.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        mlp.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)
        self.mlp = mlp

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        h = self.input_layernorm(x)
        a = self.self_attn
        q = a.q_proj(h).view(B, S, self.H, self.D)
        k = a.k_proj(h).view(B, S, self.Hk, self.D)
        v = a.v_proj(h).view(B, S, self.Hk, self.D)
        # cohere's rope is interleaved-pair; de-interleave to half-split
        # order (scores invariant to a shared head-dim permutation)
        if self.use_rope:
            d2 = self.D // 2
            q = q.reshape(B, S, self.H, d2, 2).transpose(-1, -2).reshape(B, S, self.H, self.D)
            k = k.reshape(B, S, self.Hk, d2, 2).transpose(-1, -2).reshape(B, S, self.Hk, self.D)
            q, k = apply_rope_ref(q, k, cos, sin)
        if self.sliding is not None and S > self.sliding:
            i = torch.arange(S, device=x.device)
            allowed = (i[None, :] <= i[:, None]) & (i[None, :] > i[:, None] - self.sliding)
            mask = torch.where(allowed, 0.0, float("-inf")).to(q.dtype)
            o = F.scaled_dot_product_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                attn_mask=mask, enable_gqa=self.H != self.Hk)
        else:
            o = F.scaled_dot_product_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                is_causal=True, enable_gqa=self.H != self.Hk)
        attn_out = a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        mlp_out = self.mlp.down_proj(
            F.silu(self.mlp.gate_proj(h)) * self.mlp.up_proj(h))
        return x + attn_out + mlp_out            # parallel residual


class CohereForCausalLM(nn.Module):
    hf_architectures = ("CohereForCausalLM",)
    config_class = CohereConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> CohereConfig:
        return CohereConfig.from_hf_config(hf_cfg)

    def __init__(self, config: CohereConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = CohereConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(CohereDecoderLayer(config, i)
                                     for i in range(config.num_hidden_layers))
        inner.norm = CohereLayerNorm(config.hidden_size, config.layer_norm_eps)
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
        hidden = m.norm(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            # logit_scale folds into the hidden pre-head (linear head)
            return self.loss_fn(hidden * self.config.logit_scale,
                                self.lm_head.weight, labels)
        return self.lm_head(hidden) * self.config.logit_scale

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
            elif isinstance(mod, CohereLayerNorm):
                nn.init.ones_(mod.weight)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight


class Cohere2ForCausalLM(CohereForCausalLM):
    """Cohere2 (Command-R7B): same parallel-residual block with sliding-
    window layers (rope) interleaved with NoPE full-attention layers
    (reference transformers.models.cohere2 — rope applied only when
    layer_type == "sliding_attention")."""

    hf_architectures = ("Cohere2ForCausalLM",)
