"""MiniMax (MiniMax-Text-01) causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
HYBRID layers: ``full_attention`` = GQA + rope; ``linear_attention`` =
lightning attention (silu(qkv_proj) split per head, block-wise decayed
KV-state recurrence with per-head ALiBi-style slopes, RMS-normed and
sigmoid-output-gated). Residuals are POST-LN weighted:
h = ln(x); x = h·alpha + f(h)·beta. FFN is a mixtral-class MoE
(softmax → top-k → renorm, stacked gate_up/down expert tensors). HF keys
match MiniMaxForCausalLM (parity-tested).
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
class MiniMaxConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: int | None = None
    num_local_experts: int = 8
    num_experts_per_tok: int = 2
    layer_types: list = field(default_factory=list)
    block_size: int = 256
    full_attn_alpha_factor: float = 1.0
    full_attn_beta_factor: float = 1.0
    linear_attn_alpha_factor: float = 1.0
    linear_attn_beta_factor: float = 1.0
    mlp_alpha_factor: float = 1.0
    mlp_beta_factor: float = 1.0
    max_position_embeddings: int = 131072
    rope_theta: float = 1000000.0
    rms_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "MiniMaxConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 32000),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 14336),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim"),
            num_local_experts=g("num_local_experts", 8),
            num_experts_per_tok=g("num_experts_per_tok", 2),
            layer_types=g("layer_types") or [],
            block_size=g("block_size", 256),
            full_attn_alpha_factor=g("full_attn_alpha_factor", 1.0),
            full_attn_beta_factor=g("full_attn_beta_factor", 1.0),
            linear_attn_alpha_factor=g("linear_attn_alpha_factor", 1.0),
            linear_attn_beta_factor=g("linear_attn_beta_factor", 1.0),
            mlp_alpha_factor=g("mlp_alpha_factor", 1.0),
            mlp_beta_factor=g("mlp_beta_factor", 1.0),
            max_position_embeddings=g("max_position_embeddings", 131072),
            rope_theta=rp.get("rope_theta", g("rope_theta", 1000000.0)),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class LightningAttention(nn.Module):
    """Block-decay linear attention with per-head slopes."""

    def __init__(self, cfg: MiniMaxConfig, layer_idx: int):
        super().__init__()
        H, D, E = cfg.num_attention_heads, cfg.head_dim, cfg.hidden_size
        self.H, self.D = H, D
        self.block = cfg.block_size
        self.qkv_proj = nn.Linear(E, 3 * H * D, bias=False)
        self.out_proj = nn.Linear(H * D, E, bias=False)
        self.output_gate = nn.Linear(E, H * D, bias=False)
        self.norm = RMSNorm(H * D, eps=1e-6)
        # decay schedule: slope_h = (1/2^(8/H))^(h+1) · layer-depth factor
        base = 1.0 / (2.0 ** (8.0 / H))
        factor = 1 - layer_idx / (cfg.num_hidden_layers - 1 + 1e-5) + 1e-5
        rate = (base ** torch.arange(1, H + 1).float() * factor)[:, None, None]
        r = torch.arange(self.block).float() + 1
        q_decay = torch.exp(-rate * r[:, None])
        k_decay = torch.exp(-rate * (self.block - r[:, None]))
        diag = r[:, None] - r[None, :]
        diag = rate * diag[None, None]
        diag = torch.exp(torch.where(diag >= 0, -diag, torch.tensor(float("-inf"))))
        self.register_buffer("slope_rate", rate)
        self.register_buffer("query_decay", q_decay)
        self.register_buffer("key_decay", k_decay)
        self.register_buffer("diagonal_decay", diag)

    def forward(self, x, cos=None, sin=None):
        B, S, _ = x.shape
        qkv = F.silu(self.qkv_proj(x)).view(B, S, self.H, 3 * self.D)
        q, k, v = torch.split(qkv, self.D, dim=-1)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))
        kv = x.new_zeros(B, self.H, self.D, self.D)
        outs = []
        for s0 in range(0, S, self.block):
            s1 = min(s0 + self.block, S)
            n = s1 - s0
            qb, kb, vb = q[:, :, s0:s1], k[:, :, s0:s1], v[:, :, s0:s1]
            qd = self.query_decay[:, :n]
            kd = self.key_decay[:, -n:]
            dd = self.diagonal_decay[:, :, :n, :n]
            intra = torch.matmul(torch.matmul(qb, kb.transpose(-1, -2)) * dd, vb)
            inter = torch.matmul(qb * qd, kv)
            outs.append(intra + inter)
            block_decay = torch.exp(-self.slope_rate * n)
            kv = kv * block_decay + torch.matmul((kb * kd).transpose(-1, -2), vb)
        o = torch.cat(outs, dim=-2).transpose(1, 2).reshape(B, S, self.H * self.D)
        o = self.norm(o)
        return self.out_proj(torch.sigmoid(self.output_gate(x)) * o)


class FullAttention(nn.Module):
    def __init__(self, cfg: MiniMaxConfig):
        super().__init__()
        H, Hk, D, E = (cfg.num_attention_heads, cfg.num_key_value_heads,
                       cfg.head_dim, cfg.hidden_size)
        self.H, self.Hk, self.D = H, Hk, D
        self.q_proj = nn.Linear(E, H * D, bias=False)
        self.k_proj = nn.Linear(E, Hk * D, bias=False)
        self.v_proj = nn.Linear(E, Hk * D, bias=False)
        self.o_proj = nn.Linear(H * D, E, bias=False)

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, self.H, self.D)
        k = self.k_proj(x).view(B, S, self.Hk, self.D)
        v = self.v_proj(x).view(B, S, self.Hk, self.D)
        q, k = apply_rope_ref(q, k, cos, sin)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        return self.o_proj(o.transpose(1, 2).reshape(B, S, -1))


class MiniMaxMoE(nn.Module):
    def __init__(self, cfg: MiniMaxConfig):
        super().__init__()
        E, I, H = cfg.num_local_experts, cfg.intermediate_size, cfg.hidden_size
        self.top_k = cfg.num_experts_per_tok
        gate = nn.Module()
        gate.weight = nn.Parameter(torch.empty(E, H))
        self.gate = gate
        experts = nn.Module()
        experts.gate_up_proj = nn.Parameter(torch.empty(E, 2 * I, H))
        experts.down_proj = nn.Parameter(torch.empty(E, H, I))
        self.experts = experts

    def forward(self, x):
        B, S, H = x.shape
        xf = x.reshape(-1, H)
        probs = F.softmax(F.linear(xf, self.gate.weight).float(), dim=-1)
        weights, idx = torch.topk(probs, self.top_k, dim=-1)
        weights = (weights / weights.sum(dim=-1, keepdim=True)).to(x.dtype)
        out = torch.zeros_like(xf)
        for e in idx.unique():
            tok, slot = torch.where(idx == e)
            gate, up = F.linear(xf[tok], self.experts.gate_up_proj[e]).chunk(2, dim=-1)
            h = F.silu(gate) * up
            out.index_add_(0, tok,
                           F.linear(h, self.experts.down_proj[e])
                           * weights[tok, slot, None])
        return out.view(B, S, H)


class MiniMaxLayer(nn.Module):
    def __init__(self, cfg: MiniMaxConfig, layer_idx: int):
        super().__init__()
        kinds = cfg.layer_types
        kind = kinds[layer_idx] if layer_idx < len(kinds) else "full_attention"
        self.linear = kind == "linear_attention"
        if self.linear:
            self.self_attn = LightningAttention(cfg, layer_idx)
            self.attn_alpha = cfg.linear_attn_alpha_factor
            self.attn_beta = cfg.linear_attn_beta_factor
        else:
            self.self_attn = FullAttention(cfg)
            self.attn_alpha = cfg.full_attn_alpha_factor
            self.attn_beta = cfg.full_attn_beta_factor
        self.mlp_alpha = cfg.mlp_alpha_factor
        self.mlp_beta = cfg.mlp_beta_factor
        self.input_layernorm = RMSNorm(cfg.hidden_size, eps=cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size,
                                                eps=cfg.rms_norm_eps)
        self.mlp = MiniMaxMoE(cfg)

    def forward(self, x, cos, sin):
        h = self.input_layernorm(x)      # residual taken AFTER the norm
        x = h * self.attn_alpha + self.self_attn(h, cos, sin) * self.attn_beta
        h = self.post_attention_layernorm(x)
        return h * self.mlp_alpha + self.mlp(h) * self.mlp_beta


class MiniMaxForCausalLM(nn.Module):
    hf_architectures = ("MiniMaxForCausalLM",)
    config_class = MiniMaxConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> MiniMaxConfig:
        return MiniMaxConfig.from_hf_config(hf_cfg)

    def __init__(self, config: MiniMaxConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = MiniMaxConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(MiniMaxLayer(config, i)
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
            for i, layer in enumerate(self.model.layers):
                if layer.linear:
                    fresh = LightningAttention(self.config, i)
                    for name in ("slope_rate", "query_decay", "key_decay",
                                 "diagonal_decay"):
                        getattr(layer.self_attn, name).copy_(
                            getattr(fresh, name).to(
                                layer.self_attn.slope_rate.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        for layer in self.model.layers:
            nn.init.normal_(layer.mlp.gate.weight, std=std)
            nn.init.normal_(layer.mlp.experts.gate_up_proj, std=std)
            nn.init.normal_(layer.mlp.experts.down_proj, std=std)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
