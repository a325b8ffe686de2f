"""Nemotron causal LM, MI355X-native.

Reference behavior: nemo_automodel/components/models/nemotron_v3 (NVIDIA's
own family; the reference treats it first-class). Deltas vs Llama:

  * LayerNorm-1P: LayerNorm with (1 + w) scale and a bias;
  * squared-ReLU MLP (up -> relu^2 -> down, no gate projection);
  * PARTIAL rotary: only the first head_dim * partial_rotary_factor dims
    rotate, the rest pass through.

Attention runs sdpa/flash via the shared q_start-aware helpers; GEMMs ride
hipBLASLt. HF state-dict keys match NemotronForCausalLM (parity-tested).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rope import build_rope_cache


@dataclass
class NemotronConfig:
    vocab_size: int = 256000
    hidden_size: int = 4096
    intermediate_size: int = 16384
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: int | None = None
    partial_rotary_factor: float = 0.5
    max_position_embeddings: int = 4096
    rope_theta: float = 10000.0
    norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.0134

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads

    @property
    def rms_norm_eps(self):  # recipe plumbing compatibility
        return self.norm_eps

    @classmethod
    def from_hf_config(cls, hf: Any) -> "NemotronConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 256000),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 16384),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim"),
            partial_rotary_factor=g("partial_rotary_factor", 0.5),
            max_position_embeddings=g("max_position_embeddings", 4096),
            rope_theta=g("rope_theta", 10000.0),
            norm_eps=g("norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class LayerNorm1P(nn.LayerNorm):
    """LayerNorm with zero-centered gain: y = LN(x) * (1 + w) + b."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.layer_norm(x, self.normalized_shape, self.weight + 1,
                            self.bias, self.eps)


class NemotronAttention(nn.Module):
    def __init__(self, cfg: NemotronConfig, backend: BackendConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.num_heads, self.num_kv_heads, self.head_dim = H, Hk, D
        self.rot_dim = int(D * cfg.partial_rotary_factor)
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=False)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.backend = backend

    @staticmethod
    def _rot(t: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        t1, t2 = t.chunk(2, dim=-1)
        rh = torch.cat([-t2, t1], dim=-1)
        return t * cos + rh * sin

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, -1, self.head_dim)
        k = self.k_proj(x).view(B, S, -1, self.head_dim)
        v = self.v_proj(x).view(B, S, -1, self.head_dim)
        r = self.rot_dim
        c = cos[None, :, None, :].to(q.dtype)
        s = sin[None, :, None, :].to(q.dtype)
        q = torch.cat([self._rot(q[..., :r], c, s), q[..., r:]], dim=-1)
        k = torch.cat([self._rot(k[..., :r], c, s), k[..., r:]], dim=-1)
        o = flash_attention(q, k, v, causal=True, backend="sdpa")
        return self.o_proj(o.reshape(B, S, -1))


class NemotronMLP(nn.Module):
    def __init__(self, cfg: NemotronConfig):
        super().__init__()
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down_proj(torch.relu(self.up_proj(x)).square())


class NemotronDecoderLayer(nn.Module):
    def __init__(self, cfg: NemotronConfig, backend: BackendConfig):
        super().__init__()
        self.self_attn = NemotronAttention(cfg, backend)
        self.mlp = NemotronMLP(cfg)
        self.input_layernorm = LayerNorm1P(cfg.hidden_size, eps=cfg.norm_eps)
        self.post_attention_layernorm = LayerNorm1P(cfg.hidden_size, eps=cfg.norm_eps)

    def forward(self, x, cos, sin):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class NemotronForCausalLM(nn.Module):
    hf_architectures = ("NemotronForCausalLM",)
    config_class = NemotronConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> NemotronConfig:
        return NemotronConfig.from_hf_config(hf_cfg)

    def __init__(self, config: NemotronConfig | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = NemotronConfig(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=config.head_dim)
        self.config = config
        self.backend = backend
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(NemotronDecoderLayer(config, backend)
                                     for _ in range(config.num_hidden_layers))
        inner.norm = LayerNorm1P(config.hidden_size, eps=config.norm_eps)
        rot_dim = int(config.head_dim * config.partial_rotary_factor)
        cos, sin = build_rope_cache(rot_dim, config.max_position_embeddings,
                                    config.rope_theta)
        inner.register_buffer("rope_cos", cos, persistent=False)
        inner.register_buffer("rope_sin", sin, persistent=False)
        self.model = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = inner.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        m = self.model
        x = m.embed_tokens(input_ids)
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = m.rope_cos[:S], m.rope_sin[:S]
        else:
            cos, sin = m.rope_cos[position_ids[0]], m.rope_sin[position_ids[0]]
        cos, sin = cos.float(), sin.float()
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
            rot_dim = int(self.config.head_dim * self.config.partial_rotary_factor)
            cos, sin = build_rope_cache(rot_dim,
                                        self.config.max_position_embeddings,
                                        self.config.rope_theta)
            self.model.rope_cos.copy_(cos.to(self.model.rope_cos.device))
            self.model.rope_sin.copy_(sin.to(self.model.rope_sin.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
            elif isinstance(mod, LayerNorm1P):
                nn.init.zeros_(mod.weight)
                nn.init.zeros_(mod.bias)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
