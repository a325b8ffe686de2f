"""EXAONE-4 causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
POST-norm residual blocks (x + norm(attn(x)), x + norm(mlp(x)) — no
pre-norms), per-head RMS qk-norm, SwiGLU MLP, and HYBRID attention:
``sliding_attention`` layers get rope + a causal window, ``full_attention``
layers are global NoPE (no rope at all). HF keys match
Exaone4ForCausalLM (parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache


@dataclass
class Exaone4Config:
    vocab_size: int = 102400
    hidden_size: int = 4096
    intermediate_size: int = 16384
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: int = 128
    max_position_embeddings: int = 131072
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-5
    sliding_window: int | None = 4096
    layer_types: list = field(default_factory=list)  # per-layer attention kind
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Exaone4Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 102400),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 16384),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim") or g("hidden_size", 4096) // g("num_attention_heads", 32),
            max_position_embeddings=g("max_position_embeddings", 131072),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            sliding_window=g("sliding_window"),
            layer_types=g("layer_types") or [],
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class Exaone4Layer(nn.Module):
    def __init__(self, cfg: Exaone4Config, layer_idx: int):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        kinds = cfg.layer_types
        kind = kinds[layer_idx] if layer_idx < len(kinds) else "sliding_attention"
        self.is_sliding = kind == "sliding_attention"
        self.window = cfg.sliding_window
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, eps=cfg.rms_norm_eps)
        self.post_feedforward_layernorm = RMSNorm(cfg.hidden_size,
                                                  eps=cfg.rms_norm_eps)
        attn = nn.Module()
        attn.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=False)
        attn.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        attn.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        attn.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        attn.q_norm = RMSNorm(D, eps=cfg.rms_norm_eps)
        attn.k_norm = RMSNorm(D, eps=cfg.rms_norm_eps)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        mlp.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)
        self.mlp = mlp

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        a = self.self_attn
        q = a.q_norm(a.q_proj(x).view(B, S, self.H, self.D))
        k = a.k_norm(a.k_proj(x).view(B, S, self.Hk, self.D))
        v = a.v_proj(x).view(B, S, self.Hk, self.D)
        use_window = self.window is not None and self.is_sliding
        if self.window is None or self.is_sliding:
            q, k = apply_rope_ref(q, k, cos, sin)   # global layers: NoPE
        qt, kt, vt = q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2)
        if use_window:
            i = torch.arange(S, device=x.device)
            keep = (i[None, :] <= i[:, None]) \
                & (i[None, :] > i[:, None] - self.window)
            mask = torch.where(keep, 0.0, float("-inf")) \
                .to(q.dtype).reshape(1, 1, S, S)
            o = F.scaled_dot_product_attention(
                qt, kt, vt, attn_mask=mask, enable_gqa=self.H != self.Hk)
        else:
            o = F.scaled_dot_product_attention(
                qt, kt, vt, is_causal=True, enable_gqa=self.H != self.Hk)
        attn_out = a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        x = x + self.post_attention_layernorm(attn_out)      # post-norm
        mlp_out = self.mlp.down_proj(
            F.silu(self.mlp.gate_proj(x)) * self.mlp.up_proj(x))
        return x + self.post_feedforward_layernorm(mlp_out)  # post-norm


class Exaone4ForCausalLM(nn.Module):
    hf_architectures = ("Exaone4ForCausalLM",)
    config_class = Exaone4Config

    @staticmethod
    def config_from_hf(hf_cfg) -> Exaone4Config:
        return Exaone4Config.from_hf_config(hf_cfg)

    def __init__(self, config: Exaone4Config | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = Exaone4Config(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(Exaone4Layer(config, i)
                                     for i in range(config.num_hidden_layers))
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
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
