"""StarCoder2 causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Llama-shaped pre-norm blocks but with biased LayerNorms (not RMSNorm),
biased q/k/v/o, a plain tanh-GELU MLP (c_fc/c_proj, biased), full rotary,
optional sliding window, tied embeddings. HF keys match
Starcoder2ForCausalLM (parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache


@dataclass
class Starcoder2Config:
    vocab_size: int = 49152
    hidden_size: int = 3072
    intermediate_size: int = 12288
    num_hidden_layers: int = 30
    num_attention_heads: int = 24
    num_key_value_heads: int = 2
    max_position_embeddings: int = 16384
    rope_theta: float = 100000.0
    norm_epsilon: float = 1e-5
    sliding_window: int | None = None
    tie_word_embeddings: bool = True
    initializer_range: float = 0.018042

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Starcoder2Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 49152),
            hidden_size=g("hidden_size", 3072),
            intermediate_size=g("intermediate_size", 12288),
            num_hidden_layers=g("num_hidden_layers", 30),
            num_attention_heads=g("num_attention_heads", 24),
            num_key_value_heads=g("num_key_value_heads", 2),
            max_position_embeddings=g("max_position_embeddings", 16384),
            rope_theta=rp.get("rope_theta", g("rope_theta", 100000.0)),
            norm_epsilon=g("norm_epsilon", 1e-5),
            sliding_window=g("sliding_window"),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class Starcoder2Layer(nn.Module):
    def __init__(self, cfg: Starcoder2Config):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        self.window = cfg.sliding_window
        self.input_layernorm = nn.LayerNorm(cfg.hidden_size, eps=cfg.norm_epsilon)
        self.post_attention_layernorm = nn.LayerNorm(cfg.hidden_size,
                                                     eps=cfg.norm_epsilon)
        attn = nn.Module()
        attn.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=True)
        attn.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=True)
        attn.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=True)
        attn.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=True)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.c_fc = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=True)
        mlp.c_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=True)
        self.mlp = mlp

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        h = self.input_layernorm(x)
        a = self.self_attn
        q = a.q_proj(h).view(B, S, self.H, self.D)
        k = a.k_proj(h).view(B, S, self.Hk, self.D)
        v = a.v_proj(h).view(B, S, self.Hk, self.D)
        q, k = apply_rope_ref(q, k, cos, sin)
        qt, kt, vt = q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2)
        if self.window is not None:
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
        x = x + a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        h = self.mlp.c_fc(self.post_attention_layernorm(x))
        return x + self.mlp.c_proj(F.gelu(h, approximate="tanh"))


class Starcoder2ForCausalLM(nn.Module):
    hf_architectures = ("Starcoder2ForCausalLM",)
    config_class = Starcoder2Config

    @staticmethod
    def config_from_hf(hf_cfg) -> Starcoder2Config:
        return Starcoder2Config.from_hf_config(hf_cfg)

    def __init__(self, config: Starcoder2Config | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = Starcoder2Config(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(Starcoder2Layer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.norm = nn.LayerNorm(config.hidden_size, eps=config.norm_epsilon)
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
            elif isinstance(mod, nn.LayerNorm):
                nn.init.ones_(mod.weight)
                nn.init.zeros_(mod.bias)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
