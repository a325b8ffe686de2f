"""DBRX causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Blocks hold a ``norm_attn_norm`` pair (bias-free LayerNorms) around a
clip-clamped fused ``Wqkv`` (split [H·D, Hk·D, Hk·D]) with standard rope,
then a 16-expert MoE: softmax router → top-k → p-norm weight
normalization, experts stored FLATTENED ([E·I, H] w1/v1 and a
TRANSPOSED [E·I, H] w2 applied as x@w2). HF keys match DbrxForCausalLM
(parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache


@dataclass
class DbrxConfig:
    vocab_size: int = 100352
    d_model: int = 6144
    n_layers: int = 40
    n_heads: int = 48
    kv_n_heads: int = 8
    ffn_hidden_size: int = 10752
    moe_num_experts: int = 16
    moe_top_k: int = 4
    moe_normalize_expert_weights: float | None = 1.0
    clip_qkv: float | None = 8.0
    max_seq_len: int = 32768
    rope_theta: float = 500000.0
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.d_model // self.n_heads

    @property
    def hidden_size(self):
        return self.d_model

    @property
    def num_hidden_layers(self):
        return self.n_layers

    @classmethod
    def from_hf_config(cls, hf: Any) -> "DbrxConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        attn = g("attn_config") or {}
        ffn = g("ffn_config") or {}
        if hasattr(attn, "to_dict"):
            attn = attn.to_dict()
        if hasattr(ffn, "to_dict"):
            ffn = ffn.to_dict()
        return cls(
            vocab_size=g("vocab_size", 100352),
            d_model=g("d_model", 6144),
            n_layers=g("n_layers", 40),
            n_heads=g("n_heads", 48),
            kv_n_heads=attn.get("kv_n_heads", 8),
            ffn_hidden_size=ffn.get("ffn_hidden_size", 10752),
            moe_num_experts=ffn.get("moe_num_experts", 16),
            moe_top_k=ffn.get("moe_top_k", 4),
            moe_normalize_expert_weights=ffn.get("moe_normalize_expert_weights", 1.0),
            clip_qkv=attn.get("clip_qkv", 8.0),
            max_seq_len=g("max_seq_len", 32768),
            rope_theta=attn.get("rope_theta", 500000.0),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class DbrxBlock(nn.Module):
    def __init__(self, cfg: DbrxConfig):
        super().__init__()
        E, H, Hk, D = cfg.d_model, cfg.n_heads, cfg.kv_n_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        self.clip = cfg.clip_qkv
        self.n_exp, self.inter = cfg.moe_num_experts, cfg.ffn_hidden_size
        self.top_k = cfg.moe_top_k
        self.norm_p = cfg.moe_normalize_expert_weights
        nan = nn.Module()
        nan.norm_1 = nn.LayerNorm(E, bias=False)
        nan.norm_2 = nn.LayerNorm(E, bias=False)
        attn = nn.Module()
        attn.Wqkv = nn.Linear(E, (H + 2 * Hk) * D, bias=False)
        attn.out_proj = nn.Linear(H * D, E, bias=False)
        nan.attn = attn
        self.norm_attn_norm = nan
        ffn = nn.Module()
        router = nn.Module()
        router.layer = nn.Linear(E, cfg.moe_num_experts, bias=False)
        ffn.router = router
        experts = nn.Module()
        mlp = nn.Module()
        mlp.w1 = nn.Parameter(torch.empty(self.n_exp * self.inter, E))
        mlp.v1 = nn.Parameter(torch.empty(self.n_exp * self.inter, E))
        mlp.w2 = nn.Parameter(torch.empty(self.n_exp * self.inter, E))
        experts.mlp = mlp
        ffn.experts = experts
        self.ffn = ffn

    def _moe(self, x):
        T, E = x.shape
        logits = self.ffn.router.layer(x)
        probs = F.softmax(logits, dim=-1)
        weights, idx = torch.topk(probs, self.top_k, dim=-1)
        if self.norm_p is not None:
            weights = weights / torch.norm(weights, p=self.norm_p, dim=-1,
                                           keepdim=True)
        mlp = self.ffn.experts.mlp
        w1 = mlp.w1.view(self.n_exp, self.inter, E)
        v1 = mlp.v1.view(self.n_exp, self.inter, E)
        w2 = mlp.w2.view(self.n_exp, self.inter, E)
        out = torch.zeros_like(x)
        for e in idx.unique():
            tok, slot = torch.where(idx == e)
            h = F.silu(x[tok] @ w1[e].T) * (x[tok] @ v1[e].T)
            out.index_add_(0, tok, (h @ w2[e]) * weights[tok, slot, None])
        return out

    def forward(self, x, cos, sin):
        B, S, E = x.shape
        nan = self.norm_attn_norm
        h = nan.norm_1(x)
        qkv = nan.attn.Wqkv(h)
        if self.clip is not None:
            qkv = qkv.clamp(-self.clip, self.clip)
        q, k, v = qkv.split([self.H * self.D, self.Hk * self.D,
                             self.Hk * self.D], dim=-1)
        q = q.view(B, S, self.H, self.D)
        k = k.view(B, S, self.Hk, self.D)
        v = v.view(B, S, self.Hk, self.D)
        q, k = apply_rope_ref(q, k, cos, sin)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        x = x + nan.attn.out_proj(o.transpose(1, 2).reshape(B, S, E))
        h = nan.norm_2(x)
        return x + self._moe(h.reshape(-1, E)).view(B, S, E)


class DbrxForCausalLM(nn.Module):
    hf_architectures = ("DbrxForCausalLM",)
    config_class = DbrxConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> DbrxConfig:
        return DbrxConfig.from_hf_config(hf_cfg)

    def __init__(self, config: DbrxConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = DbrxConfig(**config)
        self.config = config
        t = nn.Module()
        t.wte = nn.Embedding(config.vocab_size, config.d_model)
        t.blocks = nn.ModuleList(DbrxBlock(config)
                                 for _ in range(config.n_layers))
        t.norm_f = nn.LayerNorm(config.d_model, bias=False)
        cos, sin = build_rope_cache(config.head_dim, config.max_seq_len,
                                    config.rope_theta)
        t.register_buffer("rope_cos", cos, persistent=False)
        t.register_buffer("rope_sin", sin, persistent=False)
        self.transformer = t
        self.lm_head = nn.Linear(config.d_model, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = t.wte.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None, **_: Any):
        t = self.transformer
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = t.rope_cos[:S], t.rope_sin[:S]
        else:
            cos, sin = t.rope_cos[position_ids[0]], t.rope_sin[position_ids[0]]
        cos, sin = cos.float(), sin.float()
        x = t.wte(input_ids)
        for block in t.blocks:
            x = block(x, cos, sin)
        hidden = t.norm_f(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            cos, sin = build_rope_cache(self.config.head_dim,
                                        self.config.max_seq_len,
                                        self.config.rope_theta)
            self.transformer.rope_cos.copy_(cos.to(self.transformer.rope_cos.device))
            self.transformer.rope_sin.copy_(sin.to(self.transformer.rope_sin.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
            elif isinstance(mod, nn.LayerNorm):
                nn.init.ones_(mod.weight)
        for block in self.transformer.blocks:
            mlp = block.ffn.experts.mlp
            for pname in ("w1", "v1", "w2"):
                nn.init.normal_(getattr(mlp, pname), std=std)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.transformer.wte.weight
