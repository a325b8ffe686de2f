"""BitNet-b1.58 causal LM (bf16 master-weight path), MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Llama-shaped RMS pre-norm blocks plus SUB-norms on the sublayer outputs
(``attn_sub_norm`` before o_proj, ``ffn_sub_norm`` before down_proj) —
relu² gate activation; the QAT scaffolding of the ternary-weight recipe (pair with
quantization/qat.py for fake-quant training). HF keys match
BitNetForCausalLM (parity-tested). Attention rides sdpa.
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
class BitNetConfig:
    vocab_size: int = 128256
    hidden_size: int = 2560
    intermediate_size: int = 6912
    num_hidden_layers: int = 30
    num_attention_heads: int = 20
    num_key_value_heads: int = 5
    max_position_embeddings: int = 4096
    rope_theta: float = 500000.0
    rms_norm_eps: float = 1e-5
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "BitNetConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 128256),
            hidden_size=g("hidden_size", 2560),
            intermediate_size=g("intermediate_size", 6912),
            num_hidden_layers=g("num_hidden_layers", 30),
            num_attention_heads=g("num_attention_heads", 20),
            num_key_value_heads=g("num_key_value_heads", 5),
            max_position_embeddings=g("max_position_embeddings", 4096),
            rope_theta=rp.get("rope_theta", g("rope_theta", 500000.0)),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class BitNetLayer(nn.Module):
    def __init__(self, cfg: BitNetConfig):
        super().__init__()
        H, Hk, D, E = (cfg.num_attention_heads, cfg.num_key_value_heads,
                       cfg.head_dim, cfg.hidden_size)
        self.H, self.Hk, self.D = H, Hk, D
        self.input_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        attn = nn.Module()
        attn.q_proj = nn.Linear(E, H * D, bias=False)
        attn.k_proj = nn.Linear(E, Hk * D, bias=False)
        attn.v_proj = nn.Linear(E, Hk * D, bias=False)
        attn.o_proj = nn.Linear(H * D, E, bias=False)
        attn.attn_sub_norm = RMSNorm(H * D, eps=cfg.rms_norm_eps)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.gate_proj = nn.Linear(E, cfg.intermediate_size, bias=False)
        mlp.up_proj = nn.Linear(E, cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, E, bias=False)
        mlp.ffn_sub_norm = RMSNorm(cfg.intermediate_size, eps=cfg.rms_norm_eps)
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
        o = a.attn_sub_norm(o.transpose(1, 2).reshape(B, S, -1))
        x = x + a.o_proj(o)
        h = self.post_attention_layernorm(x)
        m = self.mlp
        return x + m.down_proj(m.ffn_sub_norm(
            F.relu(m.gate_proj(h)).square() * m.up_proj(h)))


class BitNetForCausalLM(nn.Module):
    hf_architectures = ("BitNetForCausalLM",)
    config_class = BitNetConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> BitNetConfig:
        return BitNetConfig.from_hf_config(hf_cfg)

    def __init__(self, config: BitNetConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = BitNetConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(BitNetLayer(config)
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
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
