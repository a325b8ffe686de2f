"""GPT-NeoX (Pythia) causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Biased-LayerNorm blocks with a PER-HEAD fused ``query_key_value`` ([H,3D]
rows, chunked per head), PARTIAL rotary (factor 0.25, half-split), erf-GELU
MLP (dense_h_to_4h/dense_4h_to_h, biased), and optionally PARALLEL
attention+MLP residual (``use_parallel_residual``, Pythia default). HF keys
match GPTNeoXForCausalLM (parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import build_rope_cache


@dataclass
class GPTNeoXConfig:
    vocab_size: int = 50432
    hidden_size: int = 6144
    intermediate_size: int = 24576
    num_hidden_layers: int = 44
    num_attention_heads: int = 64
    partial_rotary_factor: float = 0.25
    use_parallel_residual: bool = True
    max_position_embeddings: int = 2048
    rope_theta: float = 10000.0
    layer_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @property
    def num_key_value_heads(self):
        return self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "GPTNeoXConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 50432),
            hidden_size=g("hidden_size", 6144),
            intermediate_size=g("intermediate_size", 24576),
            num_hidden_layers=g("num_hidden_layers", 44),
            num_attention_heads=g("num_attention_heads", 64),
            partial_rotary_factor=rp.get("partial_rotary_factor",
                                         g("rotary_pct", g("partial_rotary_factor", 0.25))),
            use_parallel_residual=g("use_parallel_residual", True),
            max_position_embeddings=g("max_position_embeddings", 2048),
            rope_theta=rp.get("rope_theta", g("rope_theta", g("rotary_emb_base", 10000.0))),
            layer_norm_eps=g("layer_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class GPTNeoXLayer(nn.Module):
    def __init__(self, cfg: GPTNeoXConfig):
        super().__init__()
        H, D = cfg.num_attention_heads, cfg.head_dim
        self.H, self.D = H, D
        self.rot = int(D * cfg.partial_rotary_factor)
        self.parallel = cfg.use_parallel_residual
        E = cfg.hidden_size
        self.input_layernorm = nn.LayerNorm(E, eps=cfg.layer_norm_eps)
        self.post_attention_layernorm = nn.LayerNorm(E, eps=cfg.layer_norm_eps)
        attn = nn.Module()
        attn.query_key_value = nn.Linear(E, 3 * H * D, bias=True)
        attn.dense = nn.Linear(H * D, E, bias=True)
        self.attention = attn
        mlp = nn.Module()
        mlp.dense_h_to_4h = nn.Linear(E, cfg.intermediate_size, bias=True)
        mlp.dense_4h_to_h = nn.Linear(cfg.intermediate_size, E, bias=True)
        self.mlp = mlp

    @staticmethod
    def _rot_half(t, cos, sin):
        t1, t2 = t.chunk(2, dim=-1)
        rh = torch.cat([-t2, t1], dim=-1)
        return t * cos + rh * sin

    def _attn(self, h, cos, sin):
        B, S, _ = h.shape
        # per-head fused qkv: rows are [q_h, k_h, v_h] per head
        qkv = self.attention.query_key_value(h).view(B, S, self.H, 3 * self.D)
        q, k, v = qkv.chunk(3, dim=-1)
        r = self.rot
        q = torch.cat([self._rot_half(q[..., :r], cos, sin), q[..., r:]], dim=-1)
        k = torch.cat([self._rot_half(k[..., :r], cos, sin), k[..., r:]], dim=-1)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True)
        return self.attention.dense(o.transpose(1, 2).reshape(B, S, -1))

    def forward(self, x, cos, sin):
        attn_out = self._attn(self.input_layernorm(x), cos, sin)
        if self.parallel:
            mlp_out = self.mlp.dense_4h_to_h(
                F.gelu(self.mlp.dense_h_to_4h(self.post_attention_layernorm(x))))
            return x + attn_out + mlp_out
        x = x + attn_out
        return x + self.mlp.dense_4h_to_h(
            F.gelu(self.mlp.dense_h_to_4h(self.post_attention_layernorm(x))))


class GPTNeoXForCausalLM(nn.Module):
    hf_architectures = ("GPTNeoXForCausalLM",)
    config_class = GPTNeoXConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> GPTNeoXConfig:
        return GPTNeoXConfig.from_hf_config(hf_cfg)

    def __init__(self, config: GPTNeoXConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = GPTNeoXConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_in = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(GPTNeoXLayer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.final_layer_norm = nn.LayerNorm(config.hidden_size,
                                              eps=config.layer_norm_eps)
        rot = int(config.head_dim * config.partial_rotary_factor)
        cos, sin = build_rope_cache(rot, config.max_position_embeddings,
                                    config.rope_theta)
        inner.register_buffer("rope_cos", cos, persistent=False)
        inner.register_buffer("rope_sin", sin, persistent=False)
        self.gpt_neox = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = inner.embed_in.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None, **_: Any):
        m = self.gpt_neox
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = m.rope_cos[:S], m.rope_sin[:S]
        else:
            cos, sin = m.rope_cos[position_ids[0]], m.rope_sin[position_ids[0]]
        cos = cos.float()[None, :, None, :]
        sin = sin.float()[None, :, None, :]
        x = m.embed_in(input_ids)
        for layer in m.layers:
            x = layer(x, cos, sin)
        hidden = m.final_layer_norm(x)
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
            self.gpt_neox.rope_cos.copy_(cos.to(self.gpt_neox.rope_cos.device))
            self.gpt_neox.rope_sin.copy_(sin.to(self.gpt_neox.rope_sin.device))
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
            self.lm_head.weight = self.gpt_neox.embed_in.weight
