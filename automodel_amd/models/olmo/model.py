"""OLMo v1 and OLMo-3 causal LMs, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
OLMo v1: llama-shaped blocks with NON-PARAMETRIC LayerNorms (fp32, no
weight/bias, eps 1e-5), optional qkv clamp (``clip_qkv``), SwiGLU MLP,
standard rotary. OLMo-3: olmo2-style POST-norm sublayers + FULL-width
qk-norm, HYBRID sliding/full attention layers with (possibly) per-kind
rope theta. HF keys match OlmoForCausalLM / Olmo3ForCausalLM
(parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache


class NonParametricLayerNorm(nn.Module):
    """fp32 LayerNorm with no learnable weight/bias (OLMo v1)."""

    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.shape = (dim,)
        self.eps = eps

    def forward(self, x):
        return F.layer_norm(x.float(), self.shape, None, None,
                            eps=self.eps).to(x.dtype)


@dataclass
class OlmoConfig:
    vocab_size: int = 50304
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 32
    clip_qkv: float | None = None
    max_position_embeddings: int = 2048
    rope_theta: float = 10000.0
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "OlmoConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 50304),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 11008),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads") or g("num_attention_heads", 32),
            clip_qkv=g("clip_qkv"),
            max_position_embeddings=g("max_position_embeddings", 2048),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class OlmoLayer(nn.Module):
    def __init__(self, cfg: OlmoConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        self.clip = cfg.clip_qkv
        E = cfg.hidden_size
        self.input_layernorm = NonParametricLayerNorm(E)
        self.post_attention_layernorm = NonParametricLayerNorm(E)
        attn = nn.Module()
        attn.q_proj = nn.Linear(E, H * D, bias=False)
        attn.k_proj = nn.Linear(E, Hk * D, bias=False)
        attn.v_proj = nn.Linear(E, Hk * D, bias=False)
        attn.o_proj = nn.Linear(H * D, E, bias=False)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.gate_proj = nn.Linear(E, cfg.intermediate_size, bias=False)
        mlp.up_proj = nn.Linear(E, cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, E, bias=False)
        self.mlp = mlp

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        h = self.input_layernorm(x)
        a = self.self_attn
        q, k, v = a.q_proj(h), a.k_proj(h), a.v_proj(h)
        if self.clip is not None:
            q = q.clamp(-self.clip, self.clip)
            k = k.clamp(-self.clip, self.clip)
            v = v.clamp(-self.clip, self.clip)
        q = q.view(B, S, self.H, self.D)
        k = k.view(B, S, self.Hk, self.D)
        v = v.view(B, S, self.Hk, self.D)
        q, k = apply_rope_ref(q, k, cos, sin)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        x = x + a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        h = self.post_attention_layernorm(x)
        return x + self.mlp.down_proj(
            F.silu(self.mlp.gate_proj(h)) * self.mlp.up_proj(h))


class OlmoForCausalLM(nn.Module):
    hf_architectures = ("OlmoForCausalLM",)
    config_class = OlmoConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> OlmoConfig:
        return OlmoConfig.from_hf_config(hf_cfg)

    def __init__(self, config: OlmoConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = OlmoConfig(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(OlmoLayer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.norm = NonParametricLayerNorm(config.hidden_size)
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
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight


@dataclass
class Olmo3Config:
    vocab_size: int = 100352
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 32
    max_position_embeddings: int = 65536
    rope_theta_full: float = 500000.0
    rope_theta_sliding: float = 500000.0
    rms_norm_eps: float = 1e-6
    sliding_window: int | None = 4096
    layer_types: list = field(default_factory=list)
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Olmo3Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        # olmo3 nests per-layer-kind rope params
        full = rp.get("full_attention") or rp
        slid = rp.get("sliding_attention") or rp
        return cls(
            vocab_size=g("vocab_size", 100352),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 11008),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads") or g("num_attention_heads", 32),
            max_position_embeddings=g("max_position_embeddings", 65536),
            rope_theta_full=full.get("rope_theta") or g("rope_theta", 500000.0),
            rope_theta_sliding=slid.get("rope_theta") or g("rope_theta", 500000.0),
            rms_norm_eps=g("rms_norm_eps", 1e-6),
            sliding_window=g("sliding_window"),
            layer_types=g("layer_types") or [],
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class Olmo3Layer(nn.Module):
    def __init__(self, cfg: Olmo3Config, layer_idx: int):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.H, self.Hk, self.D = H, Hk, D
        kinds = cfg.layer_types
        kind = kinds[layer_idx] if layer_idx < len(kinds) else "full_attention"
        self.is_sliding = kind == "sliding_attention"
        self.window = cfg.sliding_window if self.is_sliding else None
        E = cfg.hidden_size
        self.post_attention_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        self.post_feedforward_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        attn = nn.Module()
        attn.q_proj = nn.Linear(E, H * D, bias=False)
        attn.k_proj = nn.Linear(E, Hk * D, bias=False)
        attn.v_proj = nn.Linear(E, Hk * D, bias=False)
        attn.o_proj = nn.Linear(H * D, E, bias=False)
        attn.q_norm = RMSNorm(H * D, eps=cfg.rms_norm_eps)   # full-width
        attn.k_norm = RMSNorm(Hk * D, eps=cfg.rms_norm_eps)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.gate_proj = nn.Linear(E, cfg.intermediate_size, bias=False)
        mlp.up_proj = nn.Linear(E, cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, E, bias=False)
        self.mlp = mlp

    def forward(self, x, cos_f, sin_f, cos_s, sin_s):
        B, S, _ = x.shape
        a = self.self_attn
        q = a.q_norm(a.q_proj(x)).view(B, S, self.H, self.D)
        k = a.k_norm(a.k_proj(x)).view(B, S, self.Hk, self.D)
        v = a.v_proj(x).view(B, S, self.Hk, self.D)
        if self.is_sliding:
            q, k = apply_rope_ref(q, k, cos_s, sin_s)
        else:
            q, k = apply_rope_ref(q, k, cos_f, sin_f)
        qt, kt, vt = q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2)
        if self.window is not None:
            i = torch.arange(S, device=x.device)
            keep = (i[None, :] <= i[:, None]) \
                & (i[None, :] > i[:, None] - self.window)
            mask = torch.where(keep, 0.0, float("-inf")) \
                .to(q.dtype).reshape(1, 1, S, S)
            o = F.scaled_dot_product_attention(
                qt, kt, vt, attn_mask=mask, enable_gqa=self.H != self.Hk)
        else:
            o = F.scaled_dot_product_attention(
                qt, kt, vt, is_causal=True, enable_gqa=self.H != self.Hk)
        attn_out = a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        x = x + self.post_attention_layernorm(attn_out)      # post-norm
        mlp_out = self.mlp.down_proj(
            F.silu(self.mlp.gate_proj(x)) * self.mlp.up_proj(x))
        return x + self.post_feedforward_layernorm(mlp_out)


class Olmo3ForCausalLM(nn.Module):
    hf_architectures = ("Olmo3ForCausalLM",)
    config_class = Olmo3Config

    @staticmethod
    def config_from_hf(hf_cfg) -> Olmo3Config:
        return Olmo3Config.from_hf_config(hf_cfg)

    def __init__(self, config: Olmo3Config | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = Olmo3Config(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(Olmo3Layer(config, i)
                                     for i in range(config.num_hidden_layers))
        inner.norm = RMSNorm(config.hidden_size, eps=config.rms_norm_eps)
        for name, theta in (("full", config.rope_theta_full),
                            ("sliding", config.rope_theta_sliding)):
            cos, sin = build_rope_cache(config.head_dim,
                                        config.max_position_embeddings, theta)
            inner.register_buffer(f"rope_cos_{name}", cos, persistent=False)
            inner.register_buffer(f"rope_sin_{name}", sin, persistent=False)
        self.model = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = inner.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None, **_: Any):
        m = self.model
        S = input_ids.shape[1]
        if position_ids is None:
            idx = slice(None, S)
        else:
            idx = position_ids[0]
        cf, sf = m.rope_cos_full[idx].float(), m.rope_sin_full[idx].float()
        cs, ss = m.rope_cos_sliding[idx].float(), m.rope_sin_sliding[idx].float()
        x = m.embed_tokens(input_ids)
        for layer in m.layers:
            x = layer(x, cf, sf, cs, ss)
        hidden = m.norm(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            for name, theta in (("full", self.config.rope_theta_full),
                                ("sliding", self.config.rope_theta_sliding)):
                cos, sin = build_rope_cache(self.config.head_dim,
                                            self.config.max_position_embeddings,
                                            theta)
                getattr(self.model, f"rope_cos_{name}").copy_(
                    cos.to(self.model.rope_cos_full.device))
                getattr(self.model, f"rope_sin_{name}").copy_(
                    sin.to(self.model.rope_cos_full.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
