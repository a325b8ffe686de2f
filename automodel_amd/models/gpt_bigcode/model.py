"""GPTBigCode (SantaCoder/StarCoder-1) causal LM, MI355X-native.

Reference behavior: the public GPTBigCode architecture (HF
transformers.models.gpt_bigcode) — GPT-2 block layout with REAL nn.Linear
weights (no Conv1D transpose), multi-query attention (fused c_attn emits
[H*D + 2*D]: per-head q, ONE shared k/v head), learned absolute positions,
tanh-GELU MLP, biased LayerNorms, tied lm_head.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.attention import flash_attention


@dataclass
class GPTBigCodeConfig:
    vocab_size: int = 50257
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    n_inner: int | None = None
    multi_query: bool = True
    layer_norm_epsilon: float = 1e-5
    scale_attn_weights: bool = True
    max_position_embeddings: int = 1024
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "GPTBigCodeConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 50257),
            hidden_size=g("n_embd") or g("hidden_size", 768),
            num_hidden_layers=g("n_layer") or g("num_hidden_layers", 12),
            num_attention_heads=g("n_head") or g("num_attention_heads", 12),
            n_inner=g("n_inner"),
            multi_query=g("multi_query", True),
            layer_norm_epsilon=g("layer_norm_epsilon", 1e-5),
            scale_attn_weights=g("scale_attn_weights", True),
            max_position_embeddings=g("n_positions") or g("max_position_embeddings", 1024),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class BigCodeAttention(nn.Module):
    def __init__(self, cfg: GPTBigCodeConfig, backend: BackendConfig):
        super().__init__()
        H = cfg.num_attention_heads
        D = cfg.hidden_size // H
        self.n_heads, self.head_dim = H, D
        self.kv_heads = 1 if cfg.multi_query else H
        self.scale = D ** -0.5 if cfg.scale_attn_weights else 1.0
        self.c_attn = nn.Linear(cfg.hidden_size,
                                cfg.hidden_size + 2 * self.kv_heads * D)
        self.c_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.backend = backend

    def forward(self, h):
        B, S, _ = h.shape
        D = self.head_dim
        qkv = self.c_attn(h)
        q, k, v = qkv.split([self.n_heads * D, self.kv_heads * D,
                             self.kv_heads * D], dim=-1)
        q = q.view(B, S, self.n_heads, D)
        k = k.view(B, S, self.kv_heads, D)
        v = v.view(B, S, self.kv_heads, D)
        o = flash_attention(q, k, v, causal=True, scale=self.scale,
                            backend=self.backend.attn)
        return self.c_proj(o.reshape(B, S, -1))


class BigCodeMLP(nn.Module):
    def __init__(self, cfg: GPTBigCodeConfig):
        super().__init__()
        inner = cfg.n_inner or 4 * cfg.hidden_size
        self.c_fc = nn.Linear(cfg.hidden_size, inner)
        self.c_proj = nn.Linear(inner, cfg.hidden_size)

    def forward(self, x):
        return self.c_proj(F.gelu(self.c_fc(x), approximate="tanh"))


class BigCodeBlock(nn.Module):
    def __init__(self, cfg: GPTBigCodeConfig, backend: BackendConfig):
        super().__init__()
        self.ln_1 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)
        self.attn = BigCodeAttention(cfg, backend)
        self.ln_2 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)
        self.mlp = BigCodeMLP(cfg)

    def forward(self, x):
        x = x + self.attn(self.ln_1(x))
        return x + self.mlp(self.ln_2(x))


class BigCodeTransformer(nn.Module):
    def __init__(self, cfg: GPTBigCodeConfig, backend: BackendConfig):
        super().__init__()
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.wpe = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        self.h = nn.ModuleList(
            BigCodeBlock(cfg, backend) for _ in range(cfg.num_hidden_layers))
        self.ln_f = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)

    def forward(self, ids, position_ids=None):
        S = ids.shape[1]
        if position_ids is None:
            position_ids = torch.arange(S, device=ids.device)
        x = self.wte(ids) + self.wpe(position_ids)
        for block in self.h:
            x = block(x)
        return self.ln_f(x)


class GPTBigCodeForCausalLM(nn.Module):
    hf_architectures = ("GPTBigCodeForCausalLM",)
    config_class = GPTBigCodeConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> GPTBigCodeConfig:
        return GPTBigCodeConfig.from_hf_config(hf_cfg)

    def __init__(self, config: GPTBigCodeConfig | dict, backend=None):
        super().__init__()
        cfg = (config if isinstance(config, GPTBigCodeConfig)
               else GPTBigCodeConfig(**dict(config)))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.hidden_size // cfg.num_attention_heads)
        self.transformer = BigCodeTransformer(cfg, bk)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.transformer.wte.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None,
                return_hidden=False, **_):
        h = self.transformer(input_ids, position_ids)
        if return_hidden:
            return h
        if labels is not None and self.loss_fn is not None:
            return self.loss_fn(h, self.lm_head.weight, labels)
        logits = self.lm_head(h)
        if labels is not None:
            return F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1),
                ignore_index=-100, reduction="sum")
        return logits

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        cfg = self.config
        if device is not None:
            self.to_empty(device=device)
        std = cfg.initializer_range
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, nn.LayerNorm):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.transformer.wte.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
