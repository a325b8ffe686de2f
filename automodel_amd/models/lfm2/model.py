"""LFM2 (LiquidAI hybrid short-conv + attention) causal LM, MI355X-native.

Reference behavior: the public LFM2 architecture (HF
transformers.models.lfm2) — layer_types mix of "conv" (gated depthwise
short conv: in_proj -> B,C,x; conv(B*x); y = C*conv; out_proj — NO
nonlinearity) and "full_attention" (GQA with per-head q/k RMSNorm, full
rope), per-layer SwiGLU (w1/w3/w2 naming) with the 2/3-adjusted rounded FF
width, final ``embedding_norm``.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache
from automodel_amd.ops.swiglu import swiglu


def lfm2_ff_dim(intermediate_size: int, auto_adjust: bool, multiplier,
                multiple_of: int) -> int:
    if not auto_adjust:
        return intermediate_size
    inter = int(2 * intermediate_size / 3)
    if multiplier is not None:
        inter = int(multiplier * inter)
        inter = multiple_of * ((inter + multiple_of - 1) // multiple_of)
    return inter


@dataclass
class Lfm2Config:
    vocab_size: int = 65536
    hidden_size: int = 1024
    intermediate_size: int = 4096
    num_hidden_layers: int = 16
    num_attention_heads: int = 16
    num_key_value_heads: int = 8
    head_dim: int | None = None
    layer_types: list = field(default_factory=list)
    conv_kernel: int = 3
    conv_bias: bool = False
    block_auto_adjust_ff_dim: bool = True
    block_ffn_dim_multiplier: float | None = 1.0
    block_multiple_of: int = 256
    norm_eps: float = 1e-5
    rope_theta: float = 1000000.0
    rope_scaling: dict | None = None
    max_position_embeddings: int = 128000
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Lfm2Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 65536),
            hidden_size=g("hidden_size", 1024),
            intermediate_size=g("intermediate_size", 4096),
            num_hidden_layers=g("num_hidden_layers", 16),
            num_attention_heads=g("num_attention_heads", 16),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim"),
            layer_types=g("layer_types") or [],
            conv_kernel=g("conv_L_cache", 3),
            conv_bias=g("conv_bias", False),
            block_auto_adjust_ff_dim=g("block_auto_adjust_ff_dim", True),
            block_ffn_dim_multiplier=g("block_ffn_dim_multiplier", 1.0),
            block_multiple_of=g("block_multiple_of", 256),
            norm_eps=g("norm_eps", 1e-5),
            rope_theta=rp.get("rope_theta", g("rope_theta", 1000000.0)),
            rope_scaling=g("rope_scaling"),
            max_position_embeddings=g("max_position_embeddings", 128000),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class Lfm2ShortConv(nn.Module):
    """Gated short conv: h = conv(B*x) elementwise-gated by C, no activation."""

    def __init__(self, cfg: Lfm2Config):
        super().__init__()
        H = cfg.hidden_size
        self.in_proj = nn.Linear(H, 3 * H, bias=cfg.conv_bias)
        self.conv = nn.Conv1d(H, H, cfg.conv_kernel, groups=H,
                              padding=cfg.conv_kernel - 1, bias=cfg.conv_bias)
        self.out_proj = nn.Linear(H, H, bias=cfg.conv_bias)

    def forward(self, h):
        S = h.shape[1]
        Bg, Cg, x = self.in_proj(h).transpose(1, 2).chunk(3, dim=1)
        y = Cg * self.conv(Bg * x)[..., :S]
        return self.out_proj(y.transpose(1, 2))


class Lfm2Attention(nn.Module):
    def __init__(self, cfg: Lfm2Config, backend: BackendConfig):
        super().__init__()
        H, Hk = cfg.num_attention_heads, cfg.num_key_value_heads
        D = cfg.head_dim or cfg.hidden_size // H
        self.head_dim = D
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=False)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.out_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.q_layernorm = RMSNorm(D, cfg.norm_eps, "torch")
        self.k_layernorm = RMSNorm(D, cfg.norm_eps, "torch")
        self.backend = backend

    def forward(self, h, cos, sin):
        B, S, _ = h.shape
        D = self.head_dim
        q = self.q_layernorm(self.q_proj(h).view(B, S, -1, D))
        k = self.k_layernorm(self.k_proj(h).view(B, S, -1, D))
        v = self.v_proj(h).view(B, S, -1, D)
        q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        return self.out_proj(o.reshape(B, S, -1))


class Lfm2MLP(nn.Module):
    def __init__(self, cfg: Lfm2Config):
        super().__init__()
        inter = lfm2_ff_dim(cfg.intermediate_size, cfg.block_auto_adjust_ff_dim,
                            cfg.block_ffn_dim_multiplier, cfg.block_multiple_of)
        self.w1 = nn.Linear(cfg.hidden_size, inter, bias=False)
        self.w3 = nn.Linear(cfg.hidden_size, inter, bias=False)
        self.w2 = nn.Linear(inter, cfg.hidden_size, bias=False)

    def forward(self, x):
        return self.w2(swiglu(self.w1(x), self.w3(x)))


class Lfm2DecoderLayer(nn.Module):
    def __init__(self, cfg: Lfm2Config, backend: BackendConfig, layer_idx: int):
        super().__init__()
        types = cfg.layer_types or ["conv"] * cfg.num_hidden_layers
        self.is_attn = types[layer_idx] == "full_attention"
        if self.is_attn:
            self.self_attn = Lfm2Attention(cfg, backend)
        else:
            self.conv = Lfm2ShortConv(cfg)
        self.feed_forward = Lfm2MLP(cfg)
        self.operator_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps, backend.rms_norm)
        self.ffn_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps, backend.rms_norm)

    def forward(self, x, cos, sin):
        h = self.operator_norm(x)
        h = self.self_attn(h, cos, sin) if self.is_attn else self.conv(h)
        x = x + h
        return x + self.feed_forward(self.ffn_norm(x))


class Lfm2Model(nn.Module):
    def __init__(self, cfg: Lfm2Config, backend: BackendConfig):
        super().__init__()
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            Lfm2DecoderLayer(cfg, backend, i) for i in range(cfg.num_hidden_layers))
        self.embedding_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps, backend.rms_norm)
        D = cfg.head_dim or cfg.hidden_size // cfg.num_attention_heads
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
        return self.embedding_norm(x)


class Lfm2ForCausalLM(nn.Module):
    hf_architectures = ("Lfm2ForCausalLM",)
    config_class = Lfm2Config

    @staticmethod
    def config_from_hf(hf_cfg) -> Lfm2Config:
        return Lfm2Config.from_hf_config(hf_cfg)

    def __init__(self, config: Lfm2Config | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, Lfm2Config) else Lfm2Config(**dict(config))
        self.config = cfg
        bk = BackendConfig.resolve(
            backend, "cuda" if torch.cuda.is_available() else "cpu",
            head_dim=cfg.head_dim or cfg.hidden_size // cfg.num_attention_heads)
        self.model = Lfm2Model(cfg, bk)
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
            return F.cross_entropy(
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
            elif type(m).__name__ == "RMSNorm":
                nn.init.ones_(m.weight)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())


# ------------------------------------------------------------- LFM2-MoE
@dataclass
class Lfm2MoeConfig(Lfm2Config):
    """LFM2-MoE: the same conv/attention hybrid with sigmoid-routed MoE
    FFNs (aux-free expert bias, DeepSeek-style) after ``num_dense_layers``
    dense SwiGLU layers. Dense layers use ``intermediate_size`` directly
    (no 2/3 auto-adjust in the MoE variant)."""

    num_experts: int = 0
    num_experts_per_tok: int = 4
    moe_intermediate_size: int = 1792
    num_dense_layers: int = 2
    use_expert_bias: bool = True
    routed_scaling_factor: float = 1.0
    norm_topk_prob: bool = True

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Lfm2MoeConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        base = Lfm2Config.from_hf_config(hf)
        cfg = cls(**{f: getattr(base, f)
                     for f in Lfm2Config.__dataclass_fields__})
        g = hf.get
        cfg.block_auto_adjust_ff_dim = False
        cfg.num_experts = g("num_experts", 32)
        cfg.num_experts_per_tok = g("num_experts_per_tok", 4)
        cfg.moe_intermediate_size = g("moe_intermediate_size", 1792)
        cfg.num_dense_layers = g("num_dense_layers", 2)
        cfg.use_expert_bias = g("use_expert_bias", True)
        cfg.routed_scaling_factor = g("routed_scaling_factor", 1.0)
        cfg.norm_topk_prob = g("norm_topk_prob", True)
        return cfg


class Lfm2MoeStateDictAdapter:
    """HF fused experts + block-level expert_bias <-> in-tree MoE keys."""

    def from_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("feed_forward.expert_bias"):
                out[k.replace("expert_bias",
                              "gate.e_score_correction_bias")] = v
            elif k.endswith("experts.gate_up_proj"):
                gate, up = v.chunk(2, dim=1)
                out[k.replace("gate_up_proj", "gate_proj")] = gate.contiguous()
                out[k.replace("gate_up_proj", "up_proj")] = up.contiguous()
            else:
                out[k] = v
        return out

    def to_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("gate.e_score_correction_bias"):
                out[k.replace("gate.e_score_correction_bias",
                              "expert_bias")] = v
            elif k.endswith("experts.gate_proj"):
                up = sd[k.replace("gate_proj", "up_proj")]
                out[k.replace("gate_proj", "gate_up_proj")] = \
                    torch.cat([v, up], dim=1)
            elif k.endswith("experts.up_proj"):
                continue
            else:
                out[k] = v
        return out


class Lfm2MoeForCausalLM(Lfm2ForCausalLM):
    hf_architectures = ("Lfm2MoeForCausalLM",)
    config_class = Lfm2MoeConfig
    state_dict_adapter = Lfm2MoeStateDictAdapter

    @staticmethod
    def config_from_hf(hf_cfg) -> Lfm2MoeConfig:
        return Lfm2MoeConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Lfm2MoeConfig | dict, backend=None):
        if isinstance(config, dict):
            config = Lfm2MoeConfig(**config)
        super().__init__(config, backend)
        from automodel_amd.moe.config import MoEConfig
        from automodel_amd.moe.layers import MoE

        moe_cfg = MoEConfig(
            n_routed_experts=config.num_experts,
            n_shared_experts=0,
            n_activated_experts=config.num_experts_per_tok,
            moe_intermediate_size=config.moe_intermediate_size,
            score_func="sigmoid", expert_bias=config.use_expert_bias,
            norm_topk_prob=config.norm_topk_prob,
            route_scale=config.routed_scaling_factor)
        for i, layer in enumerate(self.model.layers):
            if i >= config.num_dense_layers:
                layer.feed_forward = MoE(config.hidden_size, moe_cfg)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        super().init_weights(device=device)
        from automodel_amd.moe.layers import MoE

        std = self.config.initializer_range
        for m in self.modules():
            if isinstance(m, MoE):
                m.experts.init_weights(std)
                nn.init.normal_(m.gate.weight, std=std)
                if getattr(m.gate, "e_score_correction_bias", None) is not None:
                    m.gate.e_score_correction_bias.zero_()
