"""GPT-2 causal LM, MI355X-native.

Reference behavior: nemo_automodel/components/models/gpt2.py (the reference
keeps a GPT-2 for nanogpt-style pretraining). Classic architecture: learned
absolute positions (wpe), pre-LN blocks with biased LayerNorms, fused
qkv Conv1D (c_attn — HF stores Conv1D weights TRANSPOSED [in, out]),
tanh-GELU MLP, weight-tied head. Attention rides sdpa (no rope; the flash
kernel's rope-centric tiling buys nothing at GPT-2 scale).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    n_embd: int = 768
    n_layer: int = 12
    n_head: int = 4
    n_positions: int = 1024
    layer_norm_epsilon: float = 1e-5
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    # recipe plumbing aliases
    @property
    def hidden_size(self):
        return self.n_embd

    @property
    def num_hidden_layers(self):
        return self.n_layer

    @property
    def num_attention_heads(self):
        return self.n_head

    @property
    def num_key_value_heads(self):
        return self.n_head

    @property
    def head_dim(self):
        return self.n_embd // self.n_head

    @property
    def max_position_embeddings(self):
        return self.n_positions

    @classmethod
    def from_hf_config(cls, hf: Any) -> "GPT2Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 50257),
            n_embd=g("n_embd", 768),
            n_layer=g("n_layer", 12),
            n_head=g("n_head", 12),
            n_positions=g("n_positions", 1024),
            layer_norm_epsilon=g("layer_norm_epsilon", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class Conv1D(nn.Module):
    """HF GPT-2 Conv1D: weight stored [in, out] (transposed vs Linear)."""

    def __init__(self, n_in: int, n_out: int):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(n_in, n_out))
        self.bias = nn.Parameter(torch.zeros(n_out))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x @ self.weight + self.bias


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        E = cfg.n_embd
        self.ln_1 = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon)
        self.ln_2 = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon)
        attn = nn.Module()
        attn.c_attn = Conv1D(E, 3 * E)
        attn.c_proj = Conv1D(E, E)
        self.attn = attn
        mlp = nn.Module()
        mlp.c_fc = Conv1D(E, 4 * E)
        mlp.c_proj = Conv1D(4 * E, E)
        self.mlp = mlp
        self.n_head = cfg.n_head

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, E = x.shape
        h = self.ln_1(x)
        q, k, v = self.attn.c_attn(h).split(E, dim=-1)
        q = q.view(B, S, self.n_head, -1).transpose(1, 2)
        k = k.view(B, S, self.n_head, -1).transpose(1, 2)
        v = v.view(B, S, self.n_head, -1).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        x = x + self.attn.c_proj(o.transpose(1, 2).reshape(B, S, E))
        h = self.mlp.c_fc(self.ln_2(x))
        return x + self.mlp.c_proj(F.gelu(h, approximate="tanh"))


class GPT2LMHeadModel(nn.Module):
    hf_architectures = ("GPT2LMHeadModel",)
    config_class = GPT2Config

    @staticmethod
    def config_from_hf(hf_cfg) -> GPT2Config:
        return GPT2Config.from_hf_config(hf_cfg)

    def __init__(self, config: GPT2Config | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = GPT2Config(**config)
        self.config = config
        t = nn.Module()
        t.wte = nn.Embedding(config.vocab_size, config.n_embd)
        t.wpe = nn.Embedding(config.n_positions, config.n_embd)
        t.h = nn.ModuleList(GPT2Block(config) for _ in range(config.n_layer))
        t.ln_f = nn.LayerNorm(config.n_embd, eps=config.layer_norm_epsilon)
        self.transformer = t
        self.lm_head = nn.Linear(config.n_embd, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = t.wte.weight
        self.loss_fn = None

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        t = self.transformer
        S = input_ids.shape[1]
        if position_ids is None:
            position_ids = torch.arange(S, device=input_ids.device)[None]
        x = t.wte(input_ids) + t.wpe(position_ids)
        for block in t.h:
            x = block(x)
        hidden = t.ln_f(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
            elif isinstance(mod, Conv1D):
                nn.init.normal_(mod.weight, std=std)
                nn.init.zeros_(mod.bias)
            elif isinstance(mod, nn.LayerNorm):
                nn.init.ones_(mod.weight)
                nn.init.zeros_(mod.bias)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.transformer.wte.weight
