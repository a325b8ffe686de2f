"""GLM4-MoE causal LM, MI355X-native.

Reference behavior: nemo_automodel/components/models/glm4_moe/ (DeepSeek-
style MoE routing — sigmoid scores + e_score_correction_bias aux-free
balancing + shared expert + dense-first layers — under plain GQA attention
with PARTIAL rotary, factor 0.5). Reuses this framework's MoE stack
(moe/layers.py Gate/MoE with score_func="sigmoid", expert_bias=True) and
the partial-rotary attention pattern (models/nemotron). HF keys match
Glm4MoeForCausalLM (parity-tested)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.deepseek_v3.model import DenseMLP
from automodel_amd.moe.config import MoEConfig
from automodel_amd.moe.layers import MoE
from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import build_rope_cache


@dataclass
class Glm4MoeConfig:
    vocab_size: int = 151552
    hidden_size: int = 4096
    intermediate_size: int = 10944
    num_hidden_layers: int = 46
    num_attention_heads: int = 96
    num_key_value_heads: int = 8
    head_dim: int = 128
    partial_rotary_factor: float = 0.5
    first_k_dense_replace: int = 1
    attention_bias: bool = False
    use_qk_norm: bool = False
    max_position_embeddings: int = 131072
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02
    moe: MoEConfig = field(default_factory=lambda: MoEConfig(
        n_routed_experts=128, n_shared_experts=1, n_activated_experts=8,
        score_func="sigmoid", expert_bias=True, norm_topk_prob=True,
        moe_intermediate_size=1408, shared_expert_intermediate_size=1408))
    hf_flavor: str = "qwen3_moe"   # stacked expert keys (adapter layout)

    def __post_init__(self):
        if isinstance(self.moe, dict):
            self.moe = MoEConfig(**self.moe)

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Glm4MoeConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 151552),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 10944),
            num_hidden_layers=g("num_hidden_layers", 46),
            num_attention_heads=g("num_attention_heads", 96),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim", 128),
            partial_rotary_factor=g("partial_rotary_factor", 0.5),
            first_k_dense_replace=g("first_k_dense_replace", 1),
            attention_bias=g("attention_bias", False),
            use_qk_norm=g("use_qk_norm", False),
            max_position_embeddings=g("max_position_embeddings", 131072),
            rope_theta=g("rope_theta", 10000.0),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
            moe=MoEConfig(
                n_routed_experts=g("n_routed_experts", 128),
                n_shared_experts=g("n_shared_experts", 1),
                n_activated_experts=g("num_experts_per_tok", 8),
                score_func="sigmoid", expert_bias=True,
                norm_topk_prob=g("norm_topk_prob", True),
                route_scale=g("routed_scaling_factor", 1.0),
                moe_intermediate_size=g("moe_intermediate_size", 1408),
                shared_expert_intermediate_size=(
                    g("moe_intermediate_size", 1408) * g("n_shared_experts", 1)),
            ),
        )


class Glm4MoeAttention(nn.Module):
    def __init__(self, cfg: Glm4MoeConfig, backend: BackendConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.num_heads, self.num_kv_heads, self.head_dim = H, Hk, D
        self.rot_dim = int(D * cfg.partial_rotary_factor)
        b = cfg.attention_bias
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        if cfg.use_qk_norm:
            self.q_norm = RMSNorm(D, cfg.rms_norm_eps, backend.rms_norm)
            self.k_norm = RMSNorm(D, cfg.rms_norm_eps, backend.rms_norm)
        self.use_qk_norm = cfg.use_qk_norm
        self.backend = backend

    @staticmethod
    def _rot(t, cos, sin):
        t1, t2 = t.chunk(2, dim=-1)
        rh = torch.cat([-t2, t1], dim=-1)
        return t * cos + rh * sin

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, -1, self.head_dim)
        k = self.k_proj(x).view(B, S, -1, self.head_dim)
        v = self.v_proj(x).view(B, S, -1, self.head_dim)
        if self.use_qk_norm:
            q, k = self.q_norm(q), self.k_norm(k)
        r = self.rot_dim
        # [S, rot] plain tables or [B, S, rot] mrope tables (glm4v_moe)
        c = (cos[None, :, None, :] if cos.dim() == 2
             else cos[:, :, None, :]).to(q.dtype)
        s = (sin[None, :, None, :] if sin.dim() == 2
             else sin[:, :, None, :]).to(q.dtype)
        q = torch.cat([self._rot(q[..., :r], c, s), q[..., r:]], dim=-1)
        k = torch.cat([self._rot(k[..., :r], c, s), k[..., r:]], dim=-1)
        o = flash_attention(q, k, v, causal=True, backend="sdpa")
        return self.o_proj(o.reshape(B, S, -1))


class Glm4MoeDecoderLayer(nn.Module):
    def __init__(self, cfg: Glm4MoeConfig, backend: BackendConfig, layer_idx: int):
        super().__init__()
        self.self_attn = Glm4MoeAttention(cfg, backend)
        if layer_idx < cfg.first_k_dense_replace:
            self.mlp = DenseMLP(cfg.hidden_size, cfg.intermediate_size)
        else:
            self.mlp = MoE(cfg.hidden_size, cfg.moe)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                       backend.rms_norm)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                                backend.rms_norm)

    def forward(self, x, cos, sin):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class Glm4MoeForCausalLM(nn.Module):
    hf_architectures = ("Glm4MoeForCausalLM",)
    config_class = Glm4MoeConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> Glm4MoeConfig:
        return Glm4MoeConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Glm4MoeConfig | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = Glm4MoeConfig(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=config.head_dim)
        self.config = config
        self.backend = backend
        self.state_dict_adapter = MoEStateDictAdapter(config)
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(
            Glm4MoeDecoderLayer(config, backend, i)
            for i in range(config.num_hidden_layers))
        inner.norm = RMSNorm(config.hidden_size, config.rms_norm_eps,
                             backend.rms_norm)
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

    def forward(self, input_ids, labels=None, position_ids=None, **_: Any):
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
            rot = int(self.config.head_dim * self.config.partial_rotary_factor)
            cos, sin = build_rope_cache(rot, self.config.max_position_embeddings,
                                        self.config.rope_theta)
            self.model.rope_cos.copy_(cos.to(self.model.rope_cos.device))
            self.model.rope_sin.copy_(sin.to(self.model.rope_sin.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        for mod in self.modules():
            if isinstance(mod, MoE):
                mod.experts.init_weights(std)
                nn.init.normal_(mod.gate.weight, std=std)
                if getattr(mod.gate, "e_score_correction_bias", None) is not None:
                    mod.gate.e_score_correction_bias.zero_()
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
