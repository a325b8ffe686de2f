"""Phimoe (Phi-3.5-MoE) causal LM, MI355X-native.

Reference behavior: the public Phimoe architecture (HF
transformers.models.phimoe) — LayerNorm (not RMSNorm) pre-norms, biased
attention option, and the SparseMixer-v2 router (arXiv 2409.12136): top-1
by argmax within a 2eps jitter threshold band, second expert from the
re-masked scores, Gumbel sampling + the Heun third-order multiplier
correction during training (custom gradient that backprops through the
masked softmax). Expert compute rides the in-tree GroupedExperts
(grouped-GEMM HIP kernels via the probs/indices interface).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.moe.experts import GroupedExperts
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rope import apply_rope, build_rope_cache


class _SparseMixerMultiplier(torch.autograd.Function):
    """Reference PhimoeMultiplier (modeling_phimoe.py:247)."""

    @staticmethod
    def forward(ctx, scores, multiplier, selected, masked_gates, mask_for_one):
        ctx.save_for_backward(multiplier, selected, masked_gates)
        return multiplier * mask_for_one

    @staticmethod
    def backward(ctx, g):
        multiplier, selected, masked_gates = ctx.saved_tensors
        g = g * multiplier
        gs = masked_gates * g.mul(-1)
        gs.scatter_add_(dim=-1, index=selected, src=g)
        return gs, None, None, None, None


def _pick(scores, masked_scores, jitter_eps, training):
    """One sparsemixer selection round on (possibly pre-masked) scores."""
    with torch.no_grad():
        maxv, maxi = masked_scores.max(dim=-1, keepdim=True)
        factor = scores.abs().clamp(min=maxv)
        thresh = ((maxv - scores) / factor) > (2 * jitter_eps)
    gates = masked_scores.masked_fill(thresh, float("-inf"))
    if training:
        sel = ((gates - torch.empty_like(
            gates, memory_format=torch.legacy_contiguous_format)
            .exponential_().log()).max(dim=-1)[1].unsqueeze(-1))
    else:
        sel = maxi
    gates = torch.softmax(gates, dim=-1)
    mult = gates.gather(dim=-1, index=sel)
    if training:
        _, gmaxi = gates.max(dim=-1, keepdim=True)
        mask_one = torch.logical_or(sel == gmaxi,
                                    torch.rand_like(mult) > 0.75)
        mask_one = torch.add(0.3333, mask_one, alpha=0.6667).type_as(gates)
        mult = _SparseMixerMultiplier.apply(scores, mult, sel, gates, mask_one)
    return mult, sel


def sparsemixer(scores, jitter_eps, training):
    m1, s1 = _pick(scores, scores, jitter_eps, training)
    masked = torch.scatter(scores, -1, s1, float("-inf"))
    m2, s2 = _pick(scores, masked, jitter_eps, training)
    return torch.cat([m1, m2], dim=-1), torch.cat([s1, s2], dim=-1)


@dataclass
class PhimoeConfig:
    vocab_size: int = 32064
    hidden_size: int = 4096
    intermediate_size: int = 6400
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    num_local_experts: int = 16
    num_experts_per_tok: int = 2
    router_jitter_noise: float = 0.01
    input_jitter_noise: float = 0.0
    attention_bias: bool = False
    lm_head_bias: bool = False
    rms_norm_eps: float = 1e-5
    rope_theta: float = 1e6
    rope_scaling: dict | None = None
    sliding_window: int | None = None
    max_position_embeddings: int = 131072
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "PhimoeConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 32064),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 6400),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 8),
            num_local_experts=g("num_local_experts", 16),
            num_experts_per_tok=g("num_experts_per_tok", 2),
            router_jitter_noise=g("router_jitter_noise", 0.01),
            input_jitter_noise=g("input_jitter_noise", 0.0),
            attention_bias=g("attention_bias", False),
            lm_head_bias=g("lm_head_bias", False),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            rope_theta=rp.get("rope_theta", g("rope_theta", 1e6)),
            rope_scaling=g("rope_scaling"),
            sliding_window=g("sliding_window"),
            max_position_embeddings=g("max_position_embeddings", 131072),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class PhimoeAttention(nn.Module):
    def __init__(self, cfg: PhimoeConfig, backend: BackendConfig):
        super().__init__()
        H, Hk = cfg.num_attention_heads, cfg.num_key_value_heads
        D = cfg.hidden_size // H
        self.head_dim = D
        b = cfg.attention_bias
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=b)
        self.backend = backend

    def forward(self, h, cos, sin):
        B, S, _ = h.shape
        D = self.head_dim
        q = self.q_proj(h).view(B, S, -1, D)
        k = self.k_proj(h).view(B, S, -1, D)
        v = self.v_proj(h).view(B, S, -1, D)
        q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class PhimoeSparseMoeBlock(nn.Module):
    def __init__(self, cfg: PhimoeConfig):
        super().__init__()
        self.router = nn.Linear(cfg.hidden_size, cfg.num_local_experts, bias=False)
        self.experts = GroupedExperts(cfg.num_local_experts, cfg.hidden_size,
                                      cfg.intermediate_size)
        self.jitter_eps = cfg.router_jitter_noise
        self.input_jitter = cfg.input_jitter_noise

    def forward(self, x):
        B, S, H = x.shape
        if self.training and self.input_jitter > 0:
            x = x * torch.empty_like(x).uniform_(1 - self.input_jitter,
                                                 1 + self.input_jitter)
        xf = x.reshape(-1, H)
        logits = self.router(xf)
        probs, indices = sparsemixer(logits, self.jitter_eps, self.training)
        return self.experts(xf, probs.to(x.dtype), indices).view(B, S, H)


class PhimoeDecoderLayer(nn.Module):
    def __init__(self, cfg: PhimoeConfig, backend: BackendConfig):
        super().__init__()
        self.self_attn = PhimoeAttention(cfg, backend)
        self.mlp = PhimoeSparseMoeBlock(cfg)
        self.input_layernorm = nn.LayerNorm(cfg.hidden_size, eps=cfg.rms_norm_eps)
        self.post_attention_layernorm = nn.LayerNorm(cfg.hidden_size,
                                                     eps=cfg.rms_norm_eps)

    def forward(self, x, cos, sin):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin)
        return x + self.mlp(self.post_attention_layernorm(x))


class PhimoeModel(nn.Module):
    def __init__(self, cfg: PhimoeConfig, backend: BackendConfig):
        super().__init__()
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            PhimoeDecoderLayer(cfg, backend) for _ in range(cfg.num_hidden_layers))
        self.norm = nn.LayerNorm(cfg.hidden_size, eps=cfg.rms_norm_eps)
        D = cfg.hidden_size // cfg.num_attention_heads
        cos, sin = build_rope_cache(D, min(cfg.max_position_embeddings, 32768),
                                    cfg.rope_theta, cfg.rope_scaling)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, ids):
        x = self.embed_tokens(ids)
        S = x.shape[1]
        cos, sin = self.rope_cos[:S].float(), self.rope_sin[:S].float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.norm(x)


class PhimoeStateDictAdapter:
    """HF fused experts.gate_up_proj [E,2I,H] <-> stacked gate/up [E,I,H]."""

    def from_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("mlp.experts.gate_up_proj"):
                gate, up = v.chunk(2, dim=1)
                out[k.replace("gate_up_proj", "gate_proj")] = gate.contiguous()
                out[k.replace("gate_up_proj", "up_proj")] = up.contiguous()
            else:
                out[k] = v
        return out

    def to_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("mlp.experts.gate_proj"):
                up = sd[k.replace("gate_proj", "up_proj")]
                out[k.replace("gate_proj", "gate_up_proj")] = torch.cat([v, up], dim=1)
            elif k.endswith("mlp.experts.up_proj"):
                continue
            else:
                out[k] = v
        return out


class PhimoeForCausalLM(nn.Module):
    hf_architectures = ("PhimoeForCausalLM",)
    config_class = PhimoeConfig
    state_dict_adapter = PhimoeStateDictAdapter

    @staticmethod
    def config_from_hf(hf_cfg) -> PhimoeConfig:
        return PhimoeConfig.from_hf_config(hf_cfg)

    def __init__(self, config: PhimoeConfig | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, PhimoeConfig) else PhimoeConfig(**dict(config))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.hidden_size // cfg.num_attention_heads)
        self.model = PhimoeModel(cfg, bk)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=cfg.lm_head_bias)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None,
                return_hidden=False, **_):
        h = self.model(input_ids)
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
            D = cfg.hidden_size // cfg.num_attention_heads
            cos, sin = build_rope_cache(D, min(cfg.max_position_embeddings, 32768),
                                        cfg.rope_theta, cfg.rope_scaling, device=device)
            self.model.rope_cos.copy_(cos)
            self.model.rope_sin.copy_(sin)
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
            elif isinstance(m, GroupedExperts):
                m.init_weights(std)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
