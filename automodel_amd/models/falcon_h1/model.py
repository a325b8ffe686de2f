"""FalconH1 (TII hybrid parallel Mamba2 + attention) causal LM, MI355X-native.

Reference behavior: the public FalconH1 architecture (HF
transformers.models.falcon_h1) — EVERY layer runs a Mamba2 mixer AND a rope
GQA attention head-to-head on the same normed input and sums them
(ssm_out/attn_out multipliers), followed by a muP-scaled SwiGLU FFN. The
muP multipliers (embedding/lm_head/key/attention_in/out/ssm_in/out/
mlp/zxbcdt) are applied at runtime exactly as the reference does. Reuses
the shared chunked-SSD Mamba2Mixer (models/nemotron_h/model.py) with
gate_mode="silu_only" (mamba_rms_norm=False default) and the per-section
zxbcdt muP scales.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.nemotron_h.model import Mamba2Mixer
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache
from automodel_amd.ops.swiglu import swiglu


@dataclass
class FalconH1Config:
    vocab_size: int = 128000
    hidden_size: int = 2048
    intermediate_size: int = 8192
    num_hidden_layers: int = 36
    num_attention_heads: int = 16
    num_key_value_heads: int = 4
    head_dim: int | None = None
    mamba_n_heads: int = 128
    mamba_d_head: int = 64
    mamba_d_state: int = 256
    mamba_n_groups: int = 1
    mamba_conv_bias: bool = True
    mamba_proj_bias: bool = False
    mamba_chunk_size: int = 256
    mamba_rms_norm: bool = False
    mamba_norm_before_gate: bool = True
    conv_kernel: int = 4
    attention_bias: bool = False
    projectors_bias: bool = False
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    rope_scaling: dict | None = None
    max_position_embeddings: int = 8192
    tie_word_embeddings: bool = False
    hidden_act: str = "silu"
    time_step_limit: tuple = (0.0, float("inf"))
    # muP multipliers
    embedding_multiplier: float = 1.0
    lm_head_multiplier: float = 1.0
    key_multiplier: float = 1.0
    attention_in_multiplier: float = 1.0
    attention_out_multiplier: float = 1.0
    ssm_in_multiplier: float = 1.0
    ssm_out_multiplier: float = 1.0
    mlp_multipliers: tuple = (1.0, 1.0)
    ssm_multipliers: tuple = (1.0, 1.0, 1.0, 1.0, 1.0)
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "FalconH1Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 128000),
            hidden_size=g("hidden_size", 2048),
            intermediate_size=g("intermediate_size", 8192),
            num_hidden_layers=g("num_hidden_layers", 36),
            num_attention_heads=g("num_attention_heads", 16),
            num_key_value_heads=g("num_key_value_heads", 4),
            head_dim=g("head_dim"),
            mamba_n_heads=g("mamba_n_heads", 128),
            mamba_d_head=g("mamba_d_head", 64),
            mamba_d_state=g("mamba_d_state", 256),
            mamba_n_groups=g("mamba_n_groups", 1),
            mamba_conv_bias=g("mamba_conv_bias", True),
            mamba_proj_bias=g("mamba_proj_bias", False),
            mamba_chunk_size=g("mamba_chunk_size", 256),
            mamba_rms_norm=g("mamba_rms_norm", False),
            mamba_norm_before_gate=g("mamba_norm_before_gate", True),
            conv_kernel=g("mamba_d_conv", 4),
            attention_bias=g("attention_bias", False),
            projectors_bias=g("projectors_bias", False),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            rope_theta=rp.get("rope_theta", g("rope_theta") or 10000.0),
            rope_scaling=g("rope_scaling"),
            max_position_embeddings=g("max_position_embeddings", 8192),
            tie_word_embeddings=g("tie_word_embeddings", False),
            hidden_act=g("hidden_act", "silu"),
            time_step_limit=tuple(g("time_step_limit") or (0.0, float("inf"))),
            embedding_multiplier=g("embedding_multiplier", 1.0),
            lm_head_multiplier=g("lm_head_multiplier", 1.0),
            key_multiplier=g("key_multiplier", 1.0),
            attention_in_multiplier=g("attention_in_multiplier", 1.0),
            attention_out_multiplier=g("attention_out_multiplier", 1.0),
            ssm_in_multiplier=g("ssm_in_multiplier", 1.0),
            ssm_out_multiplier=g("ssm_out_multiplier", 1.0),
            mlp_multipliers=tuple(g("mlp_multipliers") or (1.0, 1.0)),
            ssm_multipliers=tuple(g("ssm_multipliers") or (1.0,) * 5),
        )


class FalconH1Attention(nn.Module):
    """Rope GQA; key_multiplier scales k right after projection (muP)."""

    def __init__(self, cfg: FalconH1Config, backend: BackendConfig):
        super().__init__()
        H, Hk = cfg.num_attention_heads, cfg.num_key_value_heads
        D = cfg.head_dim or cfg.hidden_size // H
        self.head_dim = D
        self.key_multiplier = cfg.key_multiplier
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
        k = self.k_proj(h).view(B, S, -1, D) * self.key_multiplier
        v = self.v_proj(h).view(B, S, -1, D)
        q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class FalconH1MLP(nn.Module):
    """SwiGLU with muP gate/down multipliers:
    down(up(x) * silu(gate(x) * m_gate)) * m_down."""

    def __init__(self, cfg: FalconH1Config):
        super().__init__()
        b = cfg.projectors_bias
        self.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=b)
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=b)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=b)
        self.gate_multiplier, self.down_multiplier = cfg.mlp_multipliers

    def forward(self, x):
        y = swiglu(self.gate_proj(x) * self.gate_multiplier, self.up_proj(x))
        return self.down_proj(y) * self.down_multiplier


class FalconH1DecoderLayer(nn.Module):
    def __init__(self, cfg: FalconH1Config, backend: BackendConfig):
        super().__init__()
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.pre_ff_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.feed_forward = FalconH1MLP(cfg)
        self.self_attn = FalconH1Attention(cfg, backend)
        if cfg.mamba_rms_norm:
            gate_mode = ("norm_then_gate" if cfg.mamba_norm_before_gate
                         else "norm_gate")
        else:
            gate_mode = "silu_only"
        inter = cfg.mamba_n_heads * cfg.mamba_d_head
        self.mamba = Mamba2Mixer(
            cfg.hidden_size, cfg.mamba_n_heads, cfg.mamba_d_head,
            cfg.mamba_d_state, cfg.mamba_n_groups, cfg.conv_kernel,
            cfg.mamba_chunk_size, cfg.rms_norm_eps,
            use_bias=cfg.mamba_proj_bias, use_conv_bias=cfg.mamba_conv_bias,
            norm_group_size=inter // cfg.mamba_n_groups,
            time_step_limit=cfg.time_step_limit,
            gate_mode=gate_mode, in_scale=cfg.ssm_in_multiplier,
            zxbcdt_multipliers=cfg.ssm_multipliers)
        self.attention_in_multiplier = cfg.attention_in_multiplier
        self.attn_out_multiplier = cfg.attention_out_multiplier
        self.ssm_out_multiplier = cfg.ssm_out_multiplier

    def forward(self, x, cos, sin):
        h = self.input_layernorm(x)
        mam = self.mamba(h) * self.ssm_out_multiplier
        att = self.self_attn(h * self.attention_in_multiplier, cos, sin)
        x = x + mam + att * self.attn_out_multiplier
        return x + self.feed_forward(self.pre_ff_layernorm(x))


class FalconH1Model(nn.Module):
    def __init__(self, cfg: FalconH1Config, backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            FalconH1DecoderLayer(cfg, backend) for _ in range(cfg.num_hidden_layers))
        self.final_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        D = cfg.head_dim or cfg.hidden_size // cfg.num_attention_heads
        cos, sin = build_rope_cache(D, min(cfg.max_position_embeddings, 32768),
                                    cfg.rope_theta, cfg.rope_scaling)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, ids):
        x = self.embed_tokens(ids) * self.cfg.embedding_multiplier
        S = x.shape[1]
        cos, sin = self.rope_cos[:S].float(), self.rope_sin[:S].float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.final_layernorm(x)


class FalconH1ForCausalLM(nn.Module):
    hf_architectures = ("FalconH1ForCausalLM",)
    config_class = FalconH1Config

    @staticmethod
    def config_from_hf(hf_cfg) -> FalconH1Config:
        return FalconH1Config.from_hf_config(hf_cfg)

    def __init__(self, config: FalconH1Config | dict, backend=None):
        super().__init__()
        cfg = (config if isinstance(config, FalconH1Config)
               else FalconH1Config(**dict(config)))
        self.config = cfg
        bk = BackendConfig.resolve(
            backend, "cuda" if torch.cuda.is_available() else "cpu",
            head_dim=cfg.head_dim or cfg.hidden_size // cfg.num_attention_heads)
        self.model = FalconH1Model(cfg, bk)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
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
        logits = self.lm_head(h) * self.config.lm_head_multiplier
        if labels is not None:
            return torch.nn.functional.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1),
                ignore_index=-100, reduction="sum")
        return logits

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        cfg = self.config
        if device is not None:
            self.to_empty(device=device)
            D = cfg.head_dim or cfg.hidden_size // cfg.num_attention_heads
            cos, sin = build_rope_cache(D, min(cfg.max_position_embeddings, 32768),
                                        cfg.rope_theta, cfg.rope_scaling, device=device)
            self.model.rope_cos.copy_(cos)
            self.model.rope_sin.copy_(sin)
        std = cfg.initializer_range
        for m in self.modules():
            if isinstance(m, (nn.Linear, nn.Conv1d)):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif type(m).__name__ in ("RMSNorm", "GatedRMSNorm"):
                nn.init.ones_(m.weight)
            elif isinstance(m, Mamba2Mixer):
                nn.init.ones_(m.dt_bias)
                nn.init.zeros_(m.A_log)
                nn.init.ones_(m.D)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
