"""StableLM-2 causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Llama-shaped pre-norm blocks with BIASED LayerNorms (not RMSNorm), SwiGLU
MLP, PARTIAL rotary (factor 0.25), optional qkv biases. HF keys match
StableLmForCausalLM (parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import build_rope_cache


@dataclass
class StableLmConfig:
    vocab_size: int = 50304
    hidden_size: int = 2048
    intermediate_size: int = 5632
    num_hidden_layers: int = 24
    num_attention_heads: int = 32
    num_key_value_heads: int = 32
    partial_rotary_factor: float = 0.25
    use_qkv_bias: bool = False
    max_position_embeddings: int = 4096
    rope_theta: float = 10000.0
    layer_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "StableLmConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 50304),
            hidden_size=g("hidden_size", 2048),
            intermediate_size=g("intermediate_size", 5632),
            num_hidden_layers=g("num_hidden_layers", 24),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 32),
            partial_rotary_factor=g("partial_rotary_factor", 0.25),
            use_qkv_bias=g("use_qkv_bias", False),
            max_position_embeddings=g("max_position_embeddings", 4096),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            layer_norm_eps=g("layer_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class StableLmLayer(nn.Module):
    def __init__(self, cfg: StableLmConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        self.rot = int(D * cfg.partial_rotary_factor)
        self.input_layernorm = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.post_attention_layernorm = nn.LayerNorm(cfg.hidden_size,
                                                     eps=cfg.layer_norm_eps)
        b = cfg.use_qkv_bias
        attn = nn.Module()
        attn.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        attn.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        attn.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        attn.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        mlp.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)
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
        q = a.q_proj(h).view(B, S, self.H, self.D)
        k = a.k_proj(h).view(B, S, self.Hk, self.D)
        v = a.v_proj(h).view(B, S, self.Hk, self.D)
        r = self.rot
        c, s = cos[None, :, None, :], sin[None, :, None, :]
        q = torch.cat([self._rot_half(q[..., :r], c, s), q[..., r:]], dim=-1)
        k = torch.cat([self._rot_half(k[..., :r], c, s), k[..., r:]], dim=-1)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        x = x + a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        h = self.post_attention_layernorm(x)
        return x + self.mlp.down_proj(
            F.silu(self.mlp.gate_proj(h)) * self.mlp.up_proj(h))


class StableLmForCausalLM(nn.Module):
    hf_architectures = ("StableLmForCausalLM",)
    config_class = StableLmConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> StableLmConfig:
        return StableLmConfig.from_hf_config(hf_cfg)

    def __init__(self, config: StableLmConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = StableLmConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(StableLmLayer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.norm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
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
