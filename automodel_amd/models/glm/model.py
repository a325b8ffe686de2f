"""GLM-4 dense causal LM (Glm + Glm4 layouts), MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Llama-shaped blocks with biased qkv, a FUSED ``gate_up_proj`` SwiGLU MLP
(chunk → silu(gate)·up), and PARTIAL (0.5) pair-INTERLEAVED rotary
(handled with the shared-permutation trick on the rotary channels — see
models/llama rope_interleaved). Glm4 additionally wraps each sublayer
output in a sandwich norm (``post_self_attn_layernorm`` /
``post_mlp_layernorm``) before the residual add. HF keys match
GlmForCausalLM / Glm4ForCausalLM (parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import build_rope_cache


@dataclass
class GlmConfig:
    vocab_size: int = 151552
    hidden_size: int = 4096
    intermediate_size: int = 13696
    num_hidden_layers: int = 40
    num_attention_heads: int = 32
    num_key_value_heads: int = 2
    head_dim: int = 128
    partial_rotary_factor: float = 0.5
    attention_bias: bool = True
    max_position_embeddings: int = 131072
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1.5625e-07
    sandwich_norms: bool = False        # Glm4: post_self_attn/post_mlp norms
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "GlmConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        archs = g("architectures") or []
        return cls(
            vocab_size=g("vocab_size", 151552),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 13696),
            num_hidden_layers=g("num_hidden_layers", 40),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 2),
            head_dim=g("head_dim") or g("hidden_size", 4096) // g("num_attention_heads", 32),
            partial_rotary_factor=rp.get("partial_rotary_factor",
                                         g("partial_rotary_factor", 0.5)),
            attention_bias=g("attention_bias", True),
            max_position_embeddings=g("max_position_embeddings", 131072),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            rms_norm_eps=g("rms_norm_eps", 1.5625e-07),
            sandwich_norms=any("Glm4" in a for a in archs),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class GlmLayer(nn.Module):
    def __init__(self, cfg: GlmConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        self.rot = int(D * cfg.partial_rotary_factor)
        E = cfg.hidden_size
        self.input_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        self.sandwich = cfg.sandwich_norms
        if self.sandwich:
            self.post_self_attn_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
            self.post_mlp_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        b = cfg.attention_bias
        attn = nn.Module()
        attn.q_proj = nn.Linear(E, H * D, bias=b)
        attn.k_proj = nn.Linear(E, Hk * D, bias=b)
        attn.v_proj = nn.Linear(E, Hk * D, bias=b)
        attn.o_proj = nn.Linear(H * D, E, bias=False)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.gate_up_proj = nn.Linear(E, 2 * cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, E, bias=False)
        self.mlp = mlp

    @staticmethod
    def _rot_half(t, cos, sin):
        t1, t2 = t.chunk(2, dim=-1)
        rh = torch.cat([-t2, t1], dim=-1)
        return t * cos + rh * sin

    def _rope(self, t, cos, sin):
        # partial + pair-interleaved: de-interleave the rotary channels to
        # half-split order (shared permutation on q and k — scores invariant),
        # rotate, pass the tail through untouched.
        B, S, Hn, _ = t.shape
        r = self.rot
        tr = t[..., :r].reshape(B, S, Hn, r // 2, 2).transpose(-1, -2) \
            .reshape(B, S, Hn, r)
        return torch.cat([self._rot_half(tr, cos, sin), t[..., r:]], dim=-1)

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        h = self.input_layernorm(x)
        a = self.self_attn
        q = self._rope(a.q_proj(h).view(B, S, self.H, self.D), cos, sin)
        k = self._rope(a.k_proj(h).view(B, S, self.Hk, self.D), cos, sin)
        v = a.v_proj(h).view(B, S, self.Hk, self.D)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        attn_out = a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        if self.sandwich:
            attn_out = self.post_self_attn_layernorm(attn_out)
        x = x + attn_out
        h = self.post_attention_layernorm(x)
        gate, up = self.mlp.gate_up_proj(h).chunk(2, dim=-1)
        mlp_out = self.mlp.down_proj(up * F.silu(gate))
        if self.sandwich:
            mlp_out = self.post_mlp_layernorm(mlp_out)
        return x + mlp_out


class GlmForCausalLM(nn.Module):
    hf_architectures = ("GlmForCausalLM", "Glm4ForCausalLM")
    config_class = GlmConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> GlmConfig:
        return GlmConfig.from_hf_config(hf_cfg)

    def __init__(self, config: GlmConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = GlmConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(GlmLayer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.norm = RMSNorm(config.hidden_size, eps=config.rms_norm_eps)
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
        hidden = m.norm(x)
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
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
