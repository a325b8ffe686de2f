"""Bamba (IBM hybrid Mamba2 + attention) causal LM, MI355X-native.

Reference behavior: the public Bamba architecture (HF
transformers.models.bamba) — hybrid layers (Mamba2 mixer or rope GQA
attention, chosen by attn_layer_indices) each followed by a SwiGLU FFN with
its own pre-norm. Reuses the shared chunked-SSD Mamba2Mixer
(models/nemotron_h/model.py); the gated norm is FULL-dim here (Bamba's
RMSNormGated has no grouping).
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
class BambaConfig:
    vocab_size: int = 128000
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    attn_layer_indices: list = field(default_factory=list)
    mamba_n_heads: int = 128
    mamba_d_head: int = 64
    mamba_d_state: int = 128
    mamba_n_groups: int = 1
    mamba_conv_bias: bool = True
    mamba_proj_bias: bool = False
    mamba_chunk_size: int = 256
    conv_kernel: int = 4
    attention_bias: bool = False
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    rope_scaling: dict | None = None
    partial_rotary_factor: float = 0.5
    max_position_embeddings: int = 262144
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "BambaConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 128000),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 14336),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 8),
            attn_layer_indices=g("attn_layer_indices") or [],
            mamba_n_heads=g("mamba_n_heads", 128),
            mamba_d_head=g("mamba_d_head", 64),
            mamba_d_state=g("mamba_d_state", 128),
            mamba_n_groups=g("mamba_n_groups", 1),
            mamba_conv_bias=g("mamba_conv_bias", True),
            mamba_proj_bias=g("mamba_proj_bias", False),
            mamba_chunk_size=g("mamba_chunk_size", 256),
            conv_kernel=g("mamba_d_conv", 4),
            attention_bias=g("attention_bias", False),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            rope_scaling=g("rope_scaling"),
            partial_rotary_factor=rp.get("partial_rotary_factor",
                                         g("partial_rotary_factor", 0.5)),
            max_position_embeddings=g("max_position_embeddings", 262144),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class BambaAttention(nn.Module):
    def __init__(self, cfg: BambaConfig, backend: BackendConfig):
        super().__init__()
        H, Hk = cfg.num_attention_heads, cfg.num_key_value_heads
        D = cfg.hidden_size // H
        self.head_dim = D
        self.rot = int(D * cfg.partial_rotary_factor)
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
        r = self.rot
        if r < D:
            qr, kr = apply_rope(q[..., :r].contiguous(), k[..., :r].contiguous(),
                                cos, sin, backend="torch")
            q = torch.cat([qr, q[..., r:]], dim=-1)
            k = torch.cat([kr, k[..., r:]], dim=-1)
        else:
            q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class BambaMLP(nn.Module):
    def __init__(self, cfg: BambaConfig):
        super().__init__()
        self.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


class BambaDecoderLayer(nn.Module):
    def __init__(self, cfg: BambaConfig, backend: BackendConfig, layer_idx: int):
        super().__init__()
        self.is_attn = layer_idx in (cfg.attn_layer_indices or [])
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.pre_ff_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.feed_forward = BambaMLP(cfg)
        if self.is_attn:
            self.self_attn = BambaAttention(cfg, backend)
        else:
            self.mamba = Mamba2Mixer(
                cfg.hidden_size, cfg.mamba_n_heads, cfg.mamba_d_head,
                cfg.mamba_d_state, cfg.mamba_n_groups, cfg.conv_kernel,
                cfg.mamba_chunk_size, cfg.rms_norm_eps,
                use_bias=cfg.mamba_proj_bias, use_conv_bias=cfg.mamba_conv_bias,
                norm_group_size=None)   # Bamba's gated norm is full-dim

    def forward(self, x, cos, sin):
        h = self.input_layernorm(x)
        h = self.self_attn(h, cos, sin) if self.is_attn else self.mamba(h)
        x = x + h
        return x + self.feed_forward(self.pre_ff_layernorm(x))


class BambaModel(nn.Module):
    def __init__(self, cfg: BambaConfig, backend: BackendConfig):
        super().__init__()
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            BambaDecoderLayer(cfg, backend, i) for i in range(cfg.num_hidden_layers))
        self.final_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        D = cfg.hidden_size // cfg.num_attention_heads
        rot = int(D * cfg.partial_rotary_factor)
        cos, sin = build_rope_cache(rot, min(cfg.max_position_embeddings, 32768),
                                    cfg.rope_theta, cfg.rope_scaling)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, ids):
        x = self.embed_tokens(ids)
        S = x.shape[1]
        cos, sin = self.rope_cos[:S].float(), self.rope_sin[:S].float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.final_layernorm(x)


class BambaForCausalLM(nn.Module):
    hf_architectures = ("BambaForCausalLM",)
    config_class = BambaConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> BambaConfig:
        return BambaConfig.from_hf_config(hf_cfg)

    def __init__(self, config: BambaConfig | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, BambaConfig) else BambaConfig(**dict(config))
        self.config = cfg
        bk = BackendConfig.resolve(backend, "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.hidden_size // cfg.num_attention_heads)
        self.model = BambaModel(cfg, bk)
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
        logits = self.lm_head(h)
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
            D = cfg.hidden_size // cfg.num_attention_heads
            rot = int(D * cfg.partial_rotary_factor)
            cos, sin = build_rope_cache(rot, min(cfg.max_position_embeddings, 32768),
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
